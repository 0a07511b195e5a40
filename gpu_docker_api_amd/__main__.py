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
    p.add_argument(
        "--config",
        default="",
        help="YAML config file; explicit flags override it (the reference "
        "DOCUMENTS etc/config.yaml but never implemented it — docs/en.md:126)",
    )
    p.add_argument("--addr", default=None, help="listen address (host:port)")
    p.add_argument("--portRange", default=None, dest="port_range")
    p.add_argument("--logLevel", default=None, dest="log_level")
    p.add_argument(
        "--state",
        default=None,
        help="'memory' (WAL-durable, default) or 'etcd:http://host:2379'",
    )
    p.add_argument("--dataDir", default=None, dest="data_dir")
    p.add_argument("--runtime", default=None, choices=["docker", "proc", "mock"])
    p.add_argument("--inventory", default=None, choices=["auto", "amdsmi", "mock"])
    p.add_argument("--mockGpus", type=int, default=None, dest="mock_gpus")
    p.add_argument("--dockerSocket", default=None, dest="docker_socket")
    p.add_argument("--loopVolumes", action="store_true", dest="loop_volumes",
                   help="proc runtime: enforce sized volumes via loop-mounted ext4")
    p.add_argument("--cdi", action="store_true", dest="use_cdi",
                   help="docker driver: attach GPUs via CDI (amd.com/gpu=N) instead of device nodes")
    p.add_argument("--copyEngine", default=None, choices=["auto", "iouring", "tar", "python"], dest="copy_engine")
    p.add_argument("--xgmiProbe", action="store_true", dest="run_xgmi_probe",
                   help="measure the xGMI link map with the native HIP probe at startup")
    p.add_argument("--rcclSmoke", action="store_true", dest="run_rccl_smoke",
                   help="run an RCCL all-reduce smoke test at startup")
    p.add_argument("--probeCache", default=None, dest="probe_cache")
    a = p.parse_args(argv)

    # layering: built-in defaults < YAML config file < explicit flags
    defaults = {
        "addr": "0.0.0.0:2378",
        "port_range": "40000-65535",
        "log_level": "info",
        "state": "memory",
        "runtime": "docker",
        "inventory": "auto",
        "mock_gpus": 8,
        "docker_socket": "/var/run/docker.sock",
        "copy_engine": "auto",
        "probe_cache": "",
        "data_dir": "",
    }
    from_file = {}
    if a.config:
        import yaml

        with open(a.config) as f:
            loaded = yaml.safe_load(f) or {}
        if not isinstance(loaded, dict):
            raise SystemExit(f"--config {a.config}: expected a YAML mapping")
        # accept both snake_case and the flags' camelCase spellings
        alias = {
            "portRange": "port_range", "logLevel": "log_level",
            "dataDir": "data_dir", "mockGpus": "mock_gpus",
            "dockerSocket": "docker_socket", "copyEngine": "copy_engine",
            "probeCache": "probe_cache", "xgmiProbe": "run_xgmi_probe",
            "rcclSmoke": "run_rccl_smoke", "loopVolumes": "loop_volumes",
            "cdi": "use_cdi",
        }
        for k, v in loaded.items():
            from_file[alias.get(k, k)] = v
        unknown = set(from_file) - set(defaults) - {
            "use_cdi", "loop_volumes", "run_xgmi_probe", "run_rccl_smoke", "apikey"
        }
        if unknown:
            raise SystemExit(f"--config {a.config}: unknown keys {sorted(unknown)}")

    def pick(key, flag_value):
        if flag_value is not None:
            return flag_value
        if key in from_file:
            return from_file[key]
        return defaults.get(key)

    cfg = Config(
        addr=pick("addr", a.addr),
        port_range=pick("port_range", a.port_range),
        log_level=pick("log_level", a.log_level),
        state=pick("state", a.state),
        runtime=pick("runtime", a.runtime),
        inventory=pick("inventory", a.inventory),
        mock_gpus=int(pick("mock_gpus", a.mock_gpus)),
        docker_socket=pick("docker_socket", a.docker_socket),
        use_cdi=bool(a.use_cdi or from_file.get("use_cdi", False)),
        loop_volumes=bool(a.loop_volumes or from_file.get("loop_volumes", False)),
        copy_engine=pick("copy_engine", a.copy_engine),
        run_xgmi_probe=bool(a.run_xgmi_probe or from_file.get("run_xgmi_probe", False)),
        run_rccl_smoke=bool(a.run_rccl_smoke or from_file.get("run_rccl_smoke", False)),
        probe_cache=pick("probe_cache", a.probe_cache),
    )
    data_dir = pick("data_dir", a.data_dir)
    if data_dir:
        cfg.data_dir = data_dir
    if from_file.get("apikey"):
        cfg.apikey = str(from_file["apikey"])
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
