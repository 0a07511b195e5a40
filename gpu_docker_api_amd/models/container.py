"""Request DTOs for the ReplicaSet API.

Field names mirror the reference wire contract exactly
(/root/reference/internal/models/container.go:3-56, volume.go:14-24).
"""
from __future__ import annotations

from typing import List, Optional

from pydantic import BaseModel, ConfigDict, Field


class Bind(BaseModel):
    model_config = ConfigDict(populate_by_name=True)
    src: str = ""
    dest: str = ""

    def format(self) -> str:
        if not self.src or not self.dest:
            return ""
        return f"{self.src}:{self.dest}"


class ContainerRun(BaseModel):
    model_config = ConfigDict(populate_by_name=True)
    image_name: str = Field("", alias="imageName")
    replica_set_name: str = Field("", alias="replicaSetName")
    gpu_count: int = Field(0, alias="gpuCount")
    cpu_count: int = Field(0, alias="cpuCount")
    memory: str = ""
    # extension: require this much FREE HBM on every allocated GPU
    # (e.g. "200GB") — the reference can only count whole GPUs
    gpu_memory: str = Field("", alias="gpuMemory")
    binds: List[Bind] = Field(default_factory=list)
    env: List[str] = Field(default_factory=list)
    cmd: List[str] = Field(default_factory=list)
    container_ports: List[str] = Field(default_factory=list, alias="containerPorts")


class GpuPatch(BaseModel):
    model_config = ConfigDict(populate_by_name=True)
    gpu_count: int = Field(0, alias="gpuCount")


class CpuPatch(BaseModel):
    model_config = ConfigDict(populate_by_name=True)
    cpu_count: int = Field(0, alias="cpuCount")


class MemoryPatch(BaseModel):
    memory: str = ""


class VolumePatch(BaseModel):
    model_config = ConfigDict(populate_by_name=True)
    old_bind: Optional[Bind] = Field(None, alias="oldBind")
    new_bind: Optional[Bind] = Field(None, alias="newBind")


class PatchRequest(BaseModel):
    model_config = ConfigDict(populate_by_name=True)
    gpu_patch: Optional[GpuPatch] = Field(None, alias="gpuPatch")
    cpu_patch: Optional[CpuPatch] = Field(None, alias="cpuPatch")
    memory_patch: Optional[MemoryPatch] = Field(None, alias="memoryPatch")
    volume_patch: Optional[VolumePatch] = Field(None, alias="volumePatch")

    def empty(self) -> bool:
        return not any(
            (self.gpu_patch, self.cpu_patch, self.memory_patch, self.volume_patch)
        )


class RollbackRequest(BaseModel):
    model_config = ConfigDict(populate_by_name=True)
    version: int = 0
    # extension: also restore the preserved writable layer of that version
    restore_data: bool = Field(False, alias="restoreData")


class ContainerExecute(BaseModel):
    model_config = ConfigDict(populate_by_name=True)
    work_dir: str = Field("", alias="workDir")
    cmd: List[str] = Field(default_factory=list)


class ContainerCommit(BaseModel):
    model_config = ConfigDict(populate_by_name=True)
    new_image_name: str = Field("", alias="newImageName")
