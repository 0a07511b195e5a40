#!/usr/bin/env python3
"""Export the live OpenAPI spec to api/openapi.json (diffable against the
reference's api/gpu-docker-api-en.openapi.json surface)."""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from gpu_docker_api_amd.config import Config
from gpu_docker_api_amd.routers import replicaset, resource, volume
from gpu_docker_api_amd.routers.app import build_app


def main():
    # mount routers without starting subsystems: build a shell app and
    # attach routers with stub services (route shapes don't need live deps)
    import asyncio

    from gpu_docker_api_amd.routers.app import Daemon

    cfg = Config(runtime="mock", inventory="mock", data_dir="/tmp/gda-openapi")

    async def export():
        d = Daemon(cfg)
        await d.start()
        app = build_app(cfg, daemon=d)
        from gpu_docker_api_amd.routers.app import _mount

        _mount(app, d)
        spec = app.openapi()
        await d.stop()
        return spec

    spec = asyncio.run(export())
    out = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "api")
    os.makedirs(out, exist_ok=True)
    with open(os.path.join(out, "openapi.json"), "w") as f:
        json.dump(spec, f, indent=2, sort_keys=True)
    print(f"wrote {out}/openapi.json ({len(spec.get('paths', {}))} paths)")


if __name__ == "__main__":
    main()
