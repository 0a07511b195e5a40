"""Daemon configuration.

The reference has four pflag flags + one env var + hardcoded policy constants
scattered through services. Here policy is centralized and
everything is overridable via CLI flags / env / YAML file.
"""
from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Optional


@dataclass
class Config:
    # reference defaults: main.go:33-38
    addr: str = "0.0.0.0:2378"
    port_range: str = "40000-65535"
    log_level: str = "info"

    # state backend: "memory" (WAL-durable, default) or "etcd:<endpoint>"
    state: str = "memory"
    data_dir: str = field(default_factory=lambda: os.path.join(os.getcwd(), ".state"))

    # runtime driver: mock | proc | docker ; inventory: auto | amdsmi | mock
    runtime: str = "mock"
    inventory: str = "auto"
    mock_gpus: int = 8
    docker_socket: str = "/var/run/docker.sock"
    # docker driver: attach GPUs via CDI names instead of direct device nodes
    use_cdi: bool = False

    # container policy (reference hardcodes: services/replicaset.go:67-75)
    rootfs_quota: str = "30G"
    shm_size_bytes: int = 256 * 1024**3
    restart_policy: str = "unless-stopped"

    # proc runtime: enforce sized volumes via loop-mounted ext4 images
    # (opt-in: mounts persist beyond the daemon process)
    loop_volumes: bool = False

    # copy engine: auto | iouring | tar | python
    copy_engine: str = "auto"

    # how many replaced versions' writable layers to keep under merges/
    # per replicaSet (0 = unlimited). Rollback --restore-data needs the
    # target version's layer to still be retained.
    keep_merge_layers: int = 5

    # native probes at startup (GPU boxes only)
    run_xgmi_probe: bool = False
    run_rccl_smoke: bool = False
    probe_cache: str = ""  # JSON file with previous probe results

    apikey: str = field(default_factory=lambda: os.environ.get("APIKEY", ""))

    @property
    def merges_dir(self) -> str:
        return os.path.join(self.data_dir, "merges")

    @property
    def wal_path(self) -> Optional[str]:
        if self.state == "memory":
            return os.path.join(self.data_dir, "state.wal")
        return None

    @property
    def port_start(self) -> int:
        return int(self.port_range.split("-")[0])

    @property
    def port_end(self) -> int:
        return int(self.port_range.split("-")[1])
