"""Shared test builders: a full daemon wired to the mock runtime in tmp dirs."""
from __future__ import annotations

from gpu_docker_api_amd.config import Config


def make_config(tmp_path, **overrides) -> Config:
    cfg = Config(
        state="memory",
        data_dir=str(tmp_path / "state"),
        runtime="mock",
        inventory="mock",
        mock_gpus=8,
        copy_engine="python",
        port_range="40000-40099",
    )
    for k, v in overrides.items():
        setattr(cfg, k, v)
    return cfg


async def make_daemon(tmp_path, **overrides):
    from gpu_docker_api_amd.routers.app import Daemon

    d = Daemon(make_config(tmp_path, **overrides))
    await d.start()
    return d
