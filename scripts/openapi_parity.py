#!/usr/bin/env python3
"""Route/body parity proof against the reference API surface.

Compares this daemon's live route set and request-body fields against BOTH
of the reference's sources of truth:

* its OpenAPI document (/root/reference/api/gpu-docker-api-en.openapi.json)
  — the published contract (known to under-document its own Go structs);
* its Go request structs (/root/reference/internal/models/container.go:3-56,
  volume.go:14-39) — what the daemon actually accepts.

Writes ``api/OPENAPI_PARITY.md`` and exits non-zero if any reference route
or any reference body field is missing here without a recorded explanation.
Run: ``python scripts/openapi_parity.py [path-to-reference-openapi.json]``
"""
from __future__ import annotations

import json
import os
import re
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

REF_SPEC_DEFAULT = "/root/reference/api/gpu-docker-api-en.openapi.json"

# The reference's request structs, field-by-field (wire aliases), straight
# from its Go source — its OpenAPI omits several of these.
REF_GO_BODIES = {
    ("POST", "/api/v1/replicaSet"): (
        ["imageName", "replicaSetName", "gpuCount", "cpuCount", "memory",
         "binds", "env", "cmd", "containerPorts"],
        "models/container.go:3-13",
    ),
    ("POST", "/api/v1/replicaSet/{name}/commit"): (
        ["newImageName"], "models/container.go:48-50"),
    ("POST", "/api/v1/replicaSet/{name}/execute"): (
        ["workDir", "cmd"], "models/container.go:43-46"),
    ("PATCH", "/api/v1/replicaSet/{name}"): (
        ["gpuPatch", "cpuPatch", "memoryPatch", "volumePatch"],
        "models/container.go:32-37"),
    ("PATCH", "/api/v1/replicaSet/{name}/rollback"): (
        ["version"], "models/container.go:39-41"),
    ("POST", "/api/v1/volumes"): (["name", "size"], "models/volume.go:26-29"),
    ("PATCH", "/api/v1/volumes/{name}/size"): (["size"], "models/volume.go:31-33"),
}

# Fields we accept that the reference does not — each must carry a reason.
OUR_EXTENSION_FIELDS = {
    ("POST", "/api/v1/replicaSet"): {
        "gpuMemory": "extension: minimum free HBM per allocated GPU "
                     "(MI355X 288GB HBM3E awareness; reference counts whole GPUs only)",
    },
    ("PATCH", "/api/v1/replicaSet/{name}/rollback"): {
        "restoreData": "extension: also restore that version's preserved "
                       "writable layer (reference's preservation copy is a "
                       "commented-out no-op, replicaset.go:688-698)",
    },
}

# Routes we serve beyond the reference's — each must carry a reason.
OUR_EXTENSION_ROUTES = {
    ("GET", "/api/v1/replicaSet"): "list all replicaSets (reference clients must track names externally)",
    ("GET", "/api/v1/replicaSet/{name}/logs"): "captured console output (reference users must query dockerd directly)",
    ("GET", "/api/v1/replicaSet/{name}/stats"): "live cpu/memory/pids of the current version",
    ("GET", "/api/v1/volumes"): "list all volumes",
    ("GET", "/api/v1/resources/gpus/detail"): "per-GPU HBM bytes + measured xGMI adjacency (SURVEY §2.4 row 7)",
    ("POST", "/api/v1/resources/gpus/validate"): "MFMA burn-in validation of a GPU set before handing it to a tenant",
    ("GET", "/api/v1/events"): "live state-change stream (SSE)",
    ("POST", "/api/v1/admin/compact"): "explicit history compaction (etcd compaction analog)",
    ("GET", "/api/v1/images"): "runtime image store listing",
    ("PUT", "/api/v1/images/{name}"): "register a local dir as an image (the reference delegates 'pull it locally first' to dockerd)",
    ("GET", "/metrics"): "Prometheus latency/phase metrics",
    ("GET", "/ping"): "health probe (reference has it too, outside its OpenAPI: main.go:119-123)",
}

# Reference routes its OpenAPI *omits* but its router registers.
REF_UNDOCUMENTED_ROUTES = {
    ("GET", "/api/v1/resources/cpus"): "registered in routers/resource.go:11-15, absent from the reference OpenAPI",
}


def norm(path: str) -> str:
    return re.sub(r"\{[^}]+\}", "{name}", path.rstrip("/"))


def load_reference(path: str):
    spec = json.load(open(path))
    routes, bodies = set(), {}
    for p, ops in spec.get("paths", {}).items():
        for m, op in ops.items():
            if m not in ("get", "post", "put", "patch", "delete"):
                continue
            key = (m.upper(), norm(p))
            routes.add(key)
            sch = (
                op.get("requestBody", {})
                .get("content", {})
                .get("application/json", {})
                .get("schema", {})
            )
            props = list(sch.get("properties", {}).keys())
            if props:
                bodies[key] = props
    return routes, bodies


def load_ours():
    """Live route set from the app + body fields from the pydantic DTOs."""
    import asyncio

    from gpu_docker_api_amd.config import Config
    from gpu_docker_api_amd.routers.app import Daemon, _mount, build_app
    from gpu_docker_api_amd import models as M

    cfg = Config(runtime="mock", inventory="mock", data_dir="/tmp/gda-parity")

    async def collect():
        d = Daemon(cfg)
        await d.start()
        app = build_app(cfg, daemon=d)
        _mount(app, d)
        # the generated spec flattens included routers (this FastAPI wraps
        # them in _IncludedRouter objects in app.routes)
        spec = app.openapi()
        routes = set()
        for p, ops in spec.get("paths", {}).items():
            for m in ops:
                if m in ("get", "post", "put", "patch", "delete"):
                    routes.add((m.upper(), norm(p)))
        await d.stop()
        return routes

    routes = asyncio.run(collect())

    def fields(model) -> list:
        return [f.alias or n for n, f in model.model_fields.items()]

    bodies = {
        ("POST", "/api/v1/replicaSet"): fields(M.ContainerRun),
        ("POST", "/api/v1/replicaSet/{name}/commit"): fields(M.ContainerCommit),
        ("POST", "/api/v1/replicaSet/{name}/execute"): fields(M.ContainerExecute),
        ("PATCH", "/api/v1/replicaSet/{name}"): fields(M.PatchRequest),
        ("PATCH", "/api/v1/replicaSet/{name}/rollback"): fields(M.RollbackRequest),
        ("POST", "/api/v1/volumes"): fields(M.VolumeCreate),
        ("PATCH", "/api/v1/volumes/{name}/size"): fields(M.VolumeSize),
    }
    return routes, bodies


def compare(ref_path: str = REF_SPEC_DEFAULT):
    ref_routes, ref_openapi_bodies = load_reference(ref_path)
    our_routes, our_bodies = load_ours()

    problems = []
    lines = ["# OpenAPI parity report", "",
             f"Reference spec: `{ref_path}` ({len(ref_routes)} operations); "
             f"ours: live FastAPI route table ({len(our_routes)} operations).", ""]

    lines += ["## Route parity", ""]
    missing = sorted(ref_routes - our_routes)
    for m, p in missing:
        problems.append(f"missing route {m} {p}")
        lines.append(f"- **MISSING** `{m} {p}`")
    if not missing:
        lines.append(f"- all {len(ref_routes)} reference operations are served. ✔")
    for key, why in sorted(REF_UNDOCUMENTED_ROUTES.items()):
        mark = "✔ (served here)" if key in our_routes else "✘ MISSING"
        lines.append(f"- `{key[0]} {key[1]}` — {why} — {mark}")
        if key not in our_routes:
            problems.append(f"missing undocumented reference route {key}")
    lines += ["", "### Extensions (ours beyond the reference)", ""]
    known_extra = set(OUR_EXTENSION_ROUTES)
    for m, p in sorted(our_routes - ref_routes - set(REF_UNDOCUMENTED_ROUTES)):
        why = OUR_EXTENSION_ROUTES.get((m, p))
        if why:
            lines.append(f"- `{m} {p}` — {why}")
        else:
            lines.append(f"- `{m} {p}` — **UNEXPLAINED extra route**")
            problems.append(f"unexplained extra route {m} {p}")

    lines += ["", "## Request-body field parity", "",
              "Per route: reference fields from its OpenAPI **and** its Go structs "
              "(the authoritative set — the reference's OpenAPI omits fields its "
              "daemon accepts), vs the fields our DTOs accept (wire aliases).", ""]
    for key, (go_fields, cite) in sorted(REF_GO_BODIES.items()):
        ours = our_bodies.get(key, [])
        openapi_fields = ref_openapi_bodies.get(key, [])
        missing_fields = [f for f in go_fields if f not in ours]
        extra = [f for f in ours if f not in go_fields]
        lines.append(f"### `{key[0]} {key[1]}`")
        lines.append(f"- reference Go struct ({cite}): `{', '.join(go_fields)}`")
        if openapi_fields and set(openapi_fields) != set(go_fields):
            omitted = [f for f in go_fields if f not in openapi_fields]
            lines.append(
                f"- reference OpenAPI documents only `{', '.join(openapi_fields)}` "
                f"(omits `{', '.join(omitted)}` — doc rot in the reference)")
        lines.append(f"- ours: `{', '.join(ours)}`")
        for f in missing_fields:
            lines.append(f"- **MISSING field** `{f}`")
            problems.append(f"{key}: missing body field {f}")
        for f in extra:
            why = OUR_EXTENSION_FIELDS.get(key, {}).get(f)
            if why:
                lines.append(f"- extra field `{f}` — {why}")
            else:
                lines.append(f"- **UNEXPLAINED extra field** `{f}`")
                problems.append(f"{key}: unexplained extra field {f}")
        if not missing_fields and not extra:
            lines.append("- exact match. ✔")
        lines.append("")

    lines += ["## Verdict", ""]
    if problems:
        lines.append(f"**{len(problems)} unexplained difference(s):**")
        lines += [f"- {p}" for p in problems]
    else:
        lines.append("Zero unexplained differences: every reference route and "
                     "body field is served; every extra is a recorded extension.")
    return "\n".join(lines) + "\n", problems


def main():
    ref = sys.argv[1] if len(sys.argv) > 1 else REF_SPEC_DEFAULT
    report, problems = compare(ref)
    out = os.path.join(ROOT, "api", "OPENAPI_PARITY.md")
    with open(out, "w") as f:
        f.write(report)
    print(f"wrote {out}")
    if problems:
        print("PARITY FAILURES:")
        for p in problems:
            print(" -", p)
        sys.exit(1)
    print("parity: clean")


if __name__ == "__main__":
    main()
