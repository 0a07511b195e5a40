from .base import ContainerState, RuntimeDriver, VolumeState
from .mock import MockRuntime
from .proc import ProcRuntime

__all__ = ["ContainerState", "VolumeState", "RuntimeDriver", "MockRuntime", "ProcRuntime"]


def make_runtime(kind: str, **kwargs):
    """Config-selected runtime driver (the reference needs a separate compiled
    binary per backend via Go build tags — SURVEY.md §4)."""
    if kind == "mock":
        return MockRuntime(**kwargs)
    if kind == "proc":
        return ProcRuntime(**kwargs)
    if kind == "docker":
        from .docker import DockerRuntime

        return DockerRuntime(**kwargs)
    raise ValueError(f"unknown runtime {kind!r} (expected mock|proc|docker)")
