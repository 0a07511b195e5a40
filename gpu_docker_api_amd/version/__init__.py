from .maps import MergeMap, ReleasedSet, VersionMap

__all__ = ["VersionMap", "MergeMap", "ReleasedSet"]
