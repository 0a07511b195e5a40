"""Daemon entry point: ``python -m gpu_docker_api_amd [flags]``.

Flag set supersedes the reference's four pflag flags
(/root/reference/cmd/gpu-docker-api/main.go:33-38) and adds runtime/state/
inventory selection (the reference needs separately compiled binaries for
its mock flavor).
"""
from __future__ import annotations

import argparse
import logging

from .config import Config


def parse_args(argv=None) -> Config:
    p = argparse.ArgumentParser(prog="gpu-docker-api-amd")
    p.add_argument("--addr", default="0.0.0.0:2378", help="listen address (host:port)")
    p.add_argument("--portRange", default="40000-65535", dest="port_range")
    p.add_argument("--logLevel", default="info", dest="log_level")
    p.add_argument(
        "--state",
        default="memory",
        help="'memory' (WAL-durable, default) or 'etcd:http://host:2379'",
    )
    p.add_argument("--dataDir", default="", dest="data_dir")
    p.add_argument("--runtime", default="docker", choices=["docker", "proc", "mock"])
    p.add_argument("--inventory", default="auto", choices=["auto", "amdsmi", "mock"])
    p.add_argument("--mockGpus", type=int, default=8, dest="mock_gpus")
    p.add_argument("--dockerSocket", default="/var/run/docker.sock", dest="docker_socket")
    p.add_argument("--loopVolumes", action="store_true", dest="loop_volumes",
                   help="proc runtime: enforce sized volumes via loop-mounted ext4")
    p.add_argument("--cdi", action="store_true", dest="use_cdi",
                   help="docker driver: attach GPUs via CDI (amd.com/gpu=N) instead of device nodes")
    p.add_argument("--copyEngine", default="auto", choices=["auto", "iouring", "tar", "python"], dest="copy_engine")
    p.add_argument("--xgmiProbe", action="store_true", dest="run_xgmi_probe",
                   help="measure the xGMI link map with the native HIP probe at startup")
    p.add_argument("--rcclSmoke", action="store_true", dest="run_rccl_smoke",
                   help="run an RCCL all-reduce smoke test at startup")
    p.add_argument("--probeCache", default="", dest="probe_cache")
    a = p.parse_args(argv)
    cfg = Config(
        addr=a.addr,
        port_range=a.port_range,
        log_level=a.log_level,
        state=a.state,
        runtime=a.runtime,
        inventory=a.inventory,
        mock_gpus=a.mock_gpus,
        docker_socket=a.docker_socket,
        use_cdi=a.use_cdi,
        loop_volumes=a.loop_volumes,
        copy_engine=a.copy_engine,
        run_xgmi_probe=a.run_xgmi_probe,
        run_rccl_smoke=a.run_rccl_smoke,
        probe_cache=a.probe_cache,
    )
    if a.data_dir:
        cfg.data_dir = a.data_dir
    return cfg


def main(argv=None) -> None:
    cfg = parse_args(argv)
    logging.basicConfig(
        level=getattr(logging, cfg.log_level.upper(), logging.INFO),
        format="%(asctime)s %(levelname)s %(name)s: %(message)s",
    )
    import uvicorn

    from .routers.app import build_app

    host, _, port = cfg.addr.rpartition(":")
    app = build_app(cfg)
    uvicorn.run(app, host=host or "0.0.0.0", port=int(port), log_level=cfg.log_level)


if __name__ == "__main__":
    main()
