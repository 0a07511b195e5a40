"""API fuzz: arbitrary JSON bodies must always produce the {code,msg,data}
envelope with a known business code — never an unhandled 500/stack trace."""
import pytest

hypothesis = pytest.importorskip("hypothesis")
from fastapi.testclient import TestClient
from hypothesis import HealthCheck, given, settings, strategies as st

from gpu_docker_api_amd.routers.app import build_app
from gpu_docker_api_amd.routers.codes import Code
from helpers import make_config

KNOWN_CODES = {int(c) for c in Code}

json_scalars = st.one_of(
    st.none(),
    st.booleans(),
    st.integers(min_value=-(2**31), max_value=2**31),
    st.floats(allow_nan=False, allow_infinity=False),
    st.text(max_size=30),
)
json_values = st.recursive(
    json_scalars,
    lambda inner: st.one_of(
        st.lists(inner, max_size=4), st.dictionaries(st.text(max_size=10), inner, max_size=4)
    ),
    max_leaves=12,
)


@pytest.fixture(scope="module")
def client(tmp_path_factory):
    app = build_app(make_config(tmp_path_factory.mktemp("fuzz")))
    with TestClient(app) as c:
        yield c


ROUTES = [
    ("POST", "/api/v1/replicaSet"),
    ("PATCH", "/api/v1/replicaSet/fz"),
    ("PATCH", "/api/v1/replicaSet/fz/rollback"),
    ("POST", "/api/v1/replicaSet/fz/execute"),
    ("POST", "/api/v1/replicaSet/fz/commit"),
    ("POST", "/api/v1/volumes"),
    ("PATCH", "/api/v1/volumes/fz/size"),
    ("POST", "/api/v1/resources/gpus/validate"),
    ("POST", "/api/v1/admin/compact"),
    ("PUT", "/api/v1/images/fz"),
]

# GET routes with query params fuzzed separately (no body)
GET_ROUTES = [
    "/api/v1/replicaSet/fz/logs",
    "/api/v1/replicaSet/fz/stats",
    "/api/v1/replicaSet/fz/history",
    "/api/v1/images",
]


@settings(
    max_examples=120,
    deadline=None,
    suppress_health_check=[HealthCheck.function_scoped_fixture],
)
@given(route=st.sampled_from(ROUTES), body=json_values)
def test_any_body_yields_envelope(client, route, body):
    method, path = route
    r = client.request(method, path, json=body)
    assert r.status_code == 200
    payload = r.json()
    assert set(payload) >= {"code", "msg", "data"}
    assert payload["code"] in KNOWN_CODES


@settings(
    max_examples=60,
    deadline=None,
    suppress_health_check=[HealthCheck.function_scoped_fixture],
)
@given(path=st.sampled_from(GET_ROUTES), qk=st.text(max_size=8), qv=st.text(max_size=12))
def test_get_routes_with_junk_query_yield_envelope(client, path, qk, qv):
    params = {qk: qv, "tail": qv} if qk else {"tail": qv}
    r = client.get(path, params=params)
    assert r.status_code == 200
    payload = r.json()
    assert set(payload) >= {"code", "msg", "data"}
    assert payload["code"] in KNOWN_CODES
