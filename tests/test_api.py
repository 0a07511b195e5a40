"""HTTP contract tests: same paths, envelope and numeric codes as the
reference API."""
import pytest
from fastapi.testclient import TestClient

from gpu_docker_api_amd.routers.app import build_app
from helpers import make_config


@pytest.fixture
def client(tmp_path):
    app = build_app(make_config(tmp_path))
    with TestClient(app) as c:
        yield c


RUN_BODY = {
    "imageName": "ubuntu:22.04",
    "replicaSetName": "web",
    "gpuCount": 1,
    "cpuCount": 1,
    "memory": "1GB",
    "containerPorts": ["80"],
}


def test_ping(client):
    assert client.get("/ping").json() == {"message": "pong"}


def test_run_and_info_history(client):
    r = client.post("/api/v1/replicaSet", json=RUN_BODY)
    assert r.status_code == 200
    body = r.json()
    assert body["code"] == 200 and body["msg"] == "Success"
    assert body["data"]["name"] == "web-1"

    info = client.get("/api/v1/replicaSet/web").json()
    assert info["code"] == 200
    assert info["data"]["containerName"] == "web-1"
    assert info["data"]["version"] == 1

    hist = client.get("/api/v1/replicaSet/web/history").json()
    assert hist["code"] == 200
    assert [h["version"] for h in hist["data"]] == [1]


def test_run_validation_codes(client):
    cases = [
        ({**RUN_BODY, "imageName": ""}, 1001),
        ({**RUN_BODY, "replicaSetName": ""}, 1002),
        ({**RUN_BODY, "replicaSetName": "a-b"}, 1003),
        ({**RUN_BODY, "gpuCount": -1}, 1012),
        ({**RUN_BODY, "cpuCount": -1}, 1024),
        ({**RUN_BODY, "memory": "12XB"}, 1025),
    ]
    for body, code in cases:
        assert client.post("/api/v1/replicaSet", json=body).json()["code"] == code
    # duplicate name
    assert client.post("/api/v1/replicaSet", json=RUN_BODY).json()["code"] == 200
    assert client.post("/api/v1/replicaSet", json=RUN_BODY).json()["code"] == 1008
    # gpu exhausted (mock node has 8)
    big = {**RUN_BODY, "replicaSetName": "big", "gpuCount": 99}
    assert client.post("/api/v1/replicaSet", json=big).json()["code"] == 1013


def test_patch_rollback_flow(client):
    client.post("/api/v1/replicaSet", json=RUN_BODY)
    r = client.patch("/api/v1/replicaSet/web", json={"gpuPatch": {"gpuCount": 2}})
    assert r.json()["code"] == 200
    assert r.json()["data"]["containerName"] == "web-2"
    # no-op patch => 1009
    r = client.patch("/api/v1/replicaSet/web", json={"gpuPatch": {"gpuCount": 2}})
    assert r.json()["code"] == 1009
    # rollback to same version => 1022
    r = client.patch("/api/v1/replicaSet/web/rollback", json={"version": 2})
    assert r.json()["code"] == 1022
    r = client.patch("/api/v1/replicaSet/web/rollback", json={"version": 1})
    assert r.json()["code"] == 200
    assert r.json()["data"]["containerName"] == "web-3"
    hist = client.get("/api/v1/replicaSet/web/history").json()["data"]
    assert [h["version"] for h in hist] == [3, 2, 1]


def test_lifecycle_routes(client):
    client.post("/api/v1/replicaSet", json=RUN_BODY)
    for route in ("stop", "continue", "pause", "continue"):
        assert client.patch(f"/api/v1/replicaSet/web/{route}").json()["code"] == 200
    r = client.patch("/api/v1/replicaSet/web/restart")
    assert r.json()["code"] == 200
    assert r.json()["data"]["containerName"] == "web-2"
    # execute
    r = client.post(
        "/api/v1/replicaSet/web/execute",
        json={"cmd": ["sh", "-c", "echo ok"]},
    )
    assert r.json()["code"] == 200
    assert "ok" in r.json()["data"]["stdout"]
    # commit: empty image name is rejected (the reference tags "")
    r = client.post("/api/v1/replicaSet/web/commit", json={"newImageName": ""})
    assert r.json()["code"] == 1001
    r = client.post("/api/v1/replicaSet/web/commit", json={"newImageName": "img1"})
    assert r.json()["code"] == 200 and r.json()["data"]["imageName"] == "img1"
    # delete
    assert client.delete("/api/v1/replicaSet/web").json()["code"] == 200
    assert client.get("/api/v1/replicaSet/web").json()["code"] == 1016


def test_volume_routes(client):
    assert (
        client.post("/api/v1/volumes", json={"name": "", "size": "1GB"}).json()["code"]
        == 1101
    )
    assert (
        client.post("/api/v1/volumes", json={"name": "a-b", "size": "1GB"}).json()["code"]
        == 1108
    )
    assert (
        client.post("/api/v1/volumes", json={"name": "/v", "size": "1GB"}).json()["code"]
        == 1109
    )
    assert (
        client.post("/api/v1/volumes", json={"name": "v", "size": "1XB"}).json()["code"]
        == 1106
    )
    r = client.post("/api/v1/volumes", json={"name": "v", "size": "10GB"})
    assert r.json()["code"] == 200 and r.json()["data"]["name"] == "v-1"
    assert client.post("/api/v1/volumes", json={"name": "v", "size": "10GB"}).json()["code"] == 1103
    r = client.patch("/api/v1/volumes/v/size", json={"size": "20GB"})
    assert r.json()["code"] == 200 and r.json()["data"]["name"] == "v-2"
    assert client.patch("/api/v1/volumes/v/size", json={"size": "20GB"}).json()["code"] == 1105
    info = client.get("/api/v1/volumes/v").json()
    assert info["code"] == 200 and info["data"]["version"] == 2
    hist = client.get("/api/v1/volumes/v/history").json()
    assert [h["version"] for h in hist["data"]] == [2, 1]
    assert client.delete("/api/v1/volumes/v").json()["code"] == 200
    assert client.get("/api/v1/volumes/v").json()["code"] == 1110


def test_resource_routes(client):
    gpus = client.get("/api/v1/resources/gpus").json()
    assert gpus["code"] == 200
    assert len(gpus["data"]) == 8
    assert all(v == 0 for v in gpus["data"].values())
    client.post("/api/v1/replicaSet", json=RUN_BODY)
    gpus = client.get("/api/v1/resources/gpus").json()
    assert sum(gpus["data"].values()) == 1
    detail = client.get("/api/v1/resources/gpus/detail").json()["data"]
    assert len(detail["gpus"]) == 8
    assert detail["gpus"][0]["vramTotal"] == 288 * 1024**3
    assert len(detail["xgmi"]["linkGbps"]) == 8
    cpus = client.get("/api/v1/resources/cpus").json()
    assert cpus["code"] == 200 and sum(cpus["data"].values()) == 1
    ports = client.get("/api/v1/resources/ports").json()["data"]
    assert ports["AvailableCount"] == 99
    assert len(ports["UsedPortSet"]) == 1


def test_saturation_scenario(client):
    """BASELINE config #4: 8 concurrent 1-GPU replicaSets saturate the node."""
    for i in range(8):
        body = {**RUN_BODY, "replicaSetName": f"job{i}", "gpuCount": 1}
        assert client.post("/api/v1/replicaSet", json=body).json()["code"] == 200
    gpus = client.get("/api/v1/resources/gpus").json()["data"]
    assert sum(gpus.values()) == 8  # zero free
    body = {**RUN_BODY, "replicaSetName": "job9", "gpuCount": 1}
    assert client.post("/api/v1/replicaSet", json=body).json()["code"] == 1013
    client.delete("/api/v1/replicaSet/job3")
    assert client.post("/api/v1/replicaSet", json=body).json()["code"] == 200


def test_gpu_validate_endpoint_graceful_without_gpu(client):
    r = client.post("/api/v1/resources/gpus/validate", json={})
    body = r.json()
    # CPU box: either an empty report (extension loads, 0 devices) or a
    # clean 500-coded error — never a crash
    assert body["code"] in (200, 500)
    if body["code"] == 200:
        assert body["data"]["gpus"] == []


def test_apikey_auth(tmp_path):
    cfg = make_config(tmp_path, apikey="secret-token")
    app = build_app(cfg)
    with TestClient(app) as c:
        assert c.get("/ping").status_code == 200  # ping exempt
        r = c.get("/api/v1/resources/gpus")
        assert r.json()["code"] == 403
        r = c.get(
            "/api/v1/resources/gpus", headers={"Authorization": "Bearer secret-token"}
        )
        assert r.json()["code"] == 200


def test_cors_headers(client):
    r = client.options("/api/v1/resources/gpus", headers={"Origin": "http://x.test"})
    assert r.status_code == 204
    assert r.headers["access-control-allow-origin"] == "http://x.test"
    assert r.headers["access-control-allow-credentials"] == "true"


def test_metrics_endpoint(client):
    client.post("/api/v1/replicaSet", json=RUN_BODY)
    text = client.get("/metrics").text
    assert "gda_replicaset_run_count" in text
    assert "gda_replicaset_run_p50_ms" in text


def test_image_import_and_run_from_seed(tmp_path, run):
    """Extension: PUT /images/{ref} registers a local dir; a replicaSet
    created from that image materializes its rootfs from the seed."""
    from fastapi.testclient import TestClient

    from gpu_docker_api_amd.routers.app import build_app
    from helpers import make_config

    seed = tmp_path / "seed"
    seed.mkdir()
    (seed / "hello.txt").write_text("from-the-image")

    app = build_app(make_config(tmp_path, runtime="proc"))
    with TestClient(app) as client:
        r = client.put("/api/v1/images/base:v1", json={"path": str(seed)}).json()
        assert r["code"] == 200 and r["data"]["ref"] == "base:v1"
        r = client.get("/api/v1/images").json()
        assert any(i["ref"].startswith("base") for i in r["data"])
        r = client.post(
            "/api/v1/replicaSet",
            json={"imageName": "base:v1", "replicaSetName": "seeded", "gpuCount": 0,
                  "cmd": ["sleep", "30"]},
        ).json()
        assert r["code"] == 200
        r = client.post(
            "/api/v1/replicaSet/seeded/execute",
            json={"cmd": ["cat", "hello.txt"]},
        ).json()
        assert r["code"] == 200 and "from-the-image" in r["data"]["stdout"]
        # bad path rejected
        r = client.put("/api/v1/images/x", json={"path": str(tmp_path / "nope")}).json()
        assert r["code"] != 200
        client.delete("/api/v1/replicaSet/seeded")
