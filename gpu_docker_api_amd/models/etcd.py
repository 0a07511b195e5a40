"""State-store value schemas.

Top-level JSON shape is byte-compatible with the reference's etcd values
(/root/reference/internal/models/etcd.go:12-38):

    containers/<name> -> {"version", "createTime", "config", "hostConfig",
                          "networkingConfig", "platform", "containerName"}
    volumes/<name>    -> {"version", "createTime", "opt"}

``config``/``hostConfig`` are Docker-Engine-API-shaped dicts: we keep them as
plain dicts (round-tripping unknown fields untouched) and manipulate only the
fields the control plane owns, through the typed helpers below.
"""
from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

CREATE_TIME_FMT = "%Y-%m-%d %H:%M:%S"


def _now_str() -> str:
    import datetime

    return datetime.datetime.now().strftime(CREATE_TIME_FMT)


@dataclass
class ContainerSpec:
    """Versioned container spec persisted per replicaSet name."""

    version: int = 0
    create_time: str = ""
    config: Dict[str, Any] = field(default_factory=dict)
    host_config: Dict[str, Any] = field(default_factory=dict)
    networking_config: Optional[Dict[str, Any]] = None
    platform: Optional[Dict[str, Any]] = None
    container_name: str = ""

    # ---- serialization (wire/etcd shape) ----
    def to_dict(self) -> Dict[str, Any]:
        return {
            "version": self.version,
            "createTime": self.create_time,
            "config": self.config,
            "hostConfig": self.host_config,
            "networkingConfig": self.networking_config,
            "platform": self.platform,
            "containerName": self.container_name,
        }

    def serialize(self) -> str:
        return json.dumps(self.to_dict(), separators=(",", ":"))

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ContainerSpec":
        return cls(
            version=int(d.get("version", 0)),
            create_time=d.get("createTime", ""),
            config=d.get("config") or {},
            host_config=d.get("hostConfig") or {},
            networking_config=d.get("networkingConfig"),
            platform=d.get("platform"),
            container_name=d.get("containerName", ""),
        )

    @classmethod
    def deserialize(cls, s: str) -> "ContainerSpec":
        return cls.from_dict(json.loads(s))

    # ---- typed accessors for control-plane-owned fields ----
    @property
    def env(self) -> List[str]:
        return self.config.setdefault("Env", [])

    @env.setter
    def env(self, v: List[str]) -> None:
        self.config["Env"] = v

    def set_env(self, key: str, value: str) -> None:
        env = [e for e in self.env if not e.startswith(key + "=")]
        env.append(f"{key}={value}")
        self.config["Env"] = env

    def get_env(self, key: str) -> Optional[str]:
        for e in self.env:
            if e.startswith(key + "="):
                return e[len(key) + 1 :]
        return None

    @property
    def image(self) -> str:
        return self.config.get("Image", "")

    @property
    def binds(self) -> List[str]:
        return self.host_config.setdefault("Binds", [])

    @binds.setter
    def binds(self, v: List[str]) -> None:
        self.host_config["Binds"] = v

    @property
    def memory_bytes(self) -> int:
        return int(self.host_config.get("Memory") or 0)

    @memory_bytes.setter
    def memory_bytes(self, v: int) -> None:
        self.host_config["Memory"] = int(v)

    @property
    def cpuset_cpus(self) -> str:
        return self.host_config.get("CpusetCpus", "") or ""

    @cpuset_cpus.setter
    def cpuset_cpus(self, v: str) -> None:
        self.host_config["CpusetCpus"] = v

    @property
    def devices(self) -> List[Dict[str, str]]:
        return self.host_config.setdefault("Devices", [])

    @property
    def gpu_uuids(self) -> List[str]:
        """GPU set allocated to this spec (stored in env GDA_GPU_UUIDS)."""
        v = self.get_env("GDA_GPU_UUIDS")
        return [u for u in (v.split(",") if v else []) if u]

    @gpu_uuids.setter
    def gpu_uuids(self, uuids: List[str]) -> None:
        self.set_env("GDA_GPU_UUIDS", ",".join(uuids))


@dataclass
class VolumeSpec:
    """Versioned volume spec (reference: EtcdVolumeInfo, etcd.go:28-38)."""

    version: int = 0
    create_time: str = ""
    opt: Dict[str, Any] = field(default_factory=dict)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "version": self.version,
            "createTime": self.create_time,
            "opt": self.opt,
        }

    def serialize(self) -> str:
        return json.dumps(self.to_dict(), separators=(",", ":"))

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "VolumeSpec":
        return cls(
            version=int(d.get("version", 0)),
            create_time=d.get("createTime", ""),
            opt=d.get("opt") or {},
        )

    @classmethod
    def deserialize(cls, s: str) -> "VolumeSpec":
        return cls.from_dict(json.loads(s))

    @property
    def name(self) -> str:
        return self.opt.get("Name", "")

    @property
    def size(self) -> str:
        return (self.opt.get("DriverOpts") or {}).get("size", "")

    @size.setter
    def size(self, v: str) -> None:
        self.opt.setdefault("DriverOpts", {})["size"] = v


@dataclass
class HistoryItem:
    """One history entry (reference: ContainerHistoryItem, container.go:52-56)."""

    version: int
    create_time: str
    status: Dict[str, Any]

    def to_dict(self) -> Dict[str, Any]:
        return {
            "version": self.version,
            "createTime": self.create_time,
            "status": self.status,
        }
