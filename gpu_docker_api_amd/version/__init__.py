from .maps import VersionMap, MergeMap

__all__ = ["VersionMap", "MergeMap"]
