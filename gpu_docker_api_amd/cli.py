"""Operator CLI: ``python -m gpu_docker_api_amd.cli`` (or the ``gda`` alias).

The reference ships only curl examples; this is a thin client over the same
REST API, so it works against any daemon (this one or the reference's, for
the shared routes).
"""
from __future__ import annotations

import json
import os
from typing import List, Optional

import httpx
import typer

app = typer.Typer(help="MI355X GPU container control-plane client", no_args_is_help=True)

DEFAULT_ADDR = os.environ.get("GDA_ADDR", "http://127.0.0.1:2378")


def _client() -> httpx.Client:
    headers = {}
    apikey = os.environ.get("APIKEY", "")
    if apikey:
        headers["Authorization"] = f"Bearer {apikey}"
    return httpx.Client(base_url=DEFAULT_ADDR, headers=headers, timeout=120)


def _show(resp: httpx.Response) -> None:
    body = resp.json()
    if body.get("code") != 200:
        typer.secho(f"error {body.get('code')}: {body.get('msg')}", fg="red", err=True)
        if body.get("detail"):
            typer.secho(body["detail"], fg="red", err=True)
        raise typer.Exit(1)
    typer.echo(json.dumps(body.get("data"), indent=2))


@app.command()
def run(
    name: str,
    image: str = typer.Option(..., "--image", "-i"),
    gpus: int = typer.Option(0, "--gpus", "-g"),
    cpus: int = typer.Option(0, "--cpus", "-c"),
    memory: str = typer.Option("", "--memory", "-m", help="e.g. 64GB"),
    bind: List[str] = typer.Option([], "--bind", "-b", help="src:dest"),
    env: List[str] = typer.Option([], "--env", "-e"),
    port: List[str] = typer.Option([], "--port", "-p", help="container port"),
    cmd: Optional[str] = typer.Option(None, "--cmd", help="space-separated command"),
):
    """Create + start a replicaSet."""
    binds = []
    for b in bind:
        src, _, dest = b.partition(":")
        binds.append({"src": src, "dest": dest})
    body = {
        "imageName": image,
        "replicaSetName": name,
        "gpuCount": gpus,
        "cpuCount": cpus,
        "memory": memory,
        "binds": binds,
        "env": list(env),
        "cmd": cmd.split() if cmd else [],
        "containerPorts": list(port),
    }
    with _client() as c:
        _show(c.post("/api/v1/replicaSet", json=body))


@app.command()
def ps():
    """List replicaSets."""
    with _client() as c:
        _show(c.get("/api/v1/replicaSet"))


@app.command()
def info(name: str):
    with _client() as c:
        _show(c.get(f"/api/v1/replicaSet/{name}"))


@app.command()
def history(name: str):
    with _client() as c:
        _show(c.get(f"/api/v1/replicaSet/{name}/history"))


@app.command()
def patch(
    name: str,
    gpus: Optional[int] = typer.Option(None, "--gpus", "-g"),
    cpus: Optional[int] = typer.Option(None, "--cpus", "-c"),
    memory: Optional[str] = typer.Option(None, "--memory", "-m"),
):
    """Re-scale GPU/CPU/memory (rolling replacement)."""
    body = {}
    if gpus is not None:
        body["gpuPatch"] = {"gpuCount": gpus}
    if cpus is not None:
        body["cpuPatch"] = {"cpuCount": cpus}
    if memory is not None:
        body["memoryPatch"] = {"memory": memory}
    with _client() as c:
        _show(c.patch(f"/api/v1/replicaSet/{name}", json=body))


@app.command()
def rollback(name: str, version: int):
    with _client() as c:
        _show(c.patch(f"/api/v1/replicaSet/{name}/rollback", json={"version": version}))


@app.command()
def exec(name: str, command: str, workdir: str = typer.Option("", "--workdir", "-w")):
    with _client() as c:
        _show(
            c.post(
                f"/api/v1/replicaSet/{name}/execute",
                json={"cmd": command.split(), "workDir": workdir},
            )
        )


@app.command()
def logs(name: str, tail: int = typer.Option(200, "--tail", "-n")):
    """Captured console output of the current version."""
    with _client() as c:
        r = c.get(f"/api/v1/replicaSet/{name}/logs", params={"tail": str(tail)})
        body = r.json()
        if body.get("code") == 200:
            typer.echo((body.get("data") or {}).get("logs", ""), nl=False)
        else:
            _show(r)


@app.command()
def stats(name: str):
    """Live cpu/memory/pids of the current version."""
    with _client() as c:
        _show(c.get(f"/api/v1/replicaSet/{name}/stats"))


@app.command()
def commit(name: str, image: str):
    with _client() as c:
        _show(c.post(f"/api/v1/replicaSet/{name}/commit", json={"newImageName": image}))


def _lifecycle(route: str):
    def cmd(name: str):
        with _client() as c:
            _show(c.patch(f"/api/v1/replicaSet/{name}/{route}"))

    cmd.__name__ = route
    return cmd


app.command("stop")(_lifecycle("stop"))
app.command("pause")(_lifecycle("pause"))
app.command("continue")(_lifecycle("continue"))
app.command("restart")(_lifecycle("restart"))


@app.command()
def delete(name: str):
    with _client() as c:
        _show(c.delete(f"/api/v1/replicaSet/{name}"))


vol = typer.Typer(help="volumes")
app.add_typer(vol, name="volume")


@vol.command("ls")
def vol_ls():
    with _client() as c:
        _show(c.get("/api/v1/volumes"))


@vol.command("create")
def vol_create(name: str, size: str = typer.Option("", "--size", "-s")):
    with _client() as c:
        _show(c.post("/api/v1/volumes", json={"name": name, "size": size}))


@vol.command("resize")
def vol_resize(name: str, size: str):
    with _client() as c:
        _show(c.patch(f"/api/v1/volumes/{name}/size", json={"size": size}))


@vol.command("info")
def vol_info(name: str):
    with _client() as c:
        _show(c.get(f"/api/v1/volumes/{name}"))


@vol.command("delete")
def vol_delete(name: str, keep_record: bool = typer.Option(False, "--keep-record")):
    with _client() as c:
        params = {"noall": "1"} if keep_record else {}
        _show(c.delete(f"/api/v1/volumes/{name}", params=params))


img = typer.Typer(help="runtime image store")
app.add_typer(img, name="image")


@img.command("ls")
def image_ls():
    with _client() as c:
        _show(c.get("/api/v1/images"))


@img.command("import")
def image_import(ref: str, path: str):
    """Register a local directory as an image (proc/mock runtimes)."""
    with _client() as c:
        _show(c.put(f"/api/v1/images/{ref}", json={"path": path}))


res = typer.Typer(help="node resources")
app.add_typer(res, name="resources")


@res.command("gpus")
def res_gpus(detail: bool = typer.Option(False, "--detail", "-d")):
    with _client() as c:
        path = "/api/v1/resources/gpus/detail" if detail else "/api/v1/resources/gpus"
        _show(c.get(path))


@res.command("cpus")
def res_cpus():
    with _client() as c:
        _show(c.get("/api/v1/resources/cpus"))


@res.command("ports")
def res_ports():
    with _client() as c:
        _show(c.get("/api/v1/resources/ports"))


@res.command("validate")
def res_validate(
    size: int = typer.Option(4096, "--size"),
    iters: int = typer.Option(5, "--iters"),
):
    """Burn-in free GPUs (HBM bandwidth + bf16/MX-fp8/MX-fp4 MFMA TFLOPS)."""
    with _client() as c:
        _show(
            c.post(
                "/api/v1/resources/gpus/validate",
                json={"size": size, "iters": iters},
                timeout=600,
            )
        )


@app.command()
def events():
    """Tail the live state-change stream (SSE); Ctrl-C to stop."""
    headers = {}
    apikey = os.environ.get("APIKEY", "")
    if apikey:
        headers["Authorization"] = f"Bearer {apikey}"
    with httpx.stream(
        "GET", f"{DEFAULT_ADDR}/api/v1/events", headers=headers, timeout=None
    ) as resp:
        for line in resp.iter_lines():
            if line.startswith("data: "):
                typer.echo(line[6:])


def main() -> None:
    app()


if __name__ == "__main__":
    main()
