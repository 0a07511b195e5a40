"""Property-based MVCC semantics: random operation sequences against a
simple oracle (full snapshots per revision). The reference's rollback
correctness rests entirely on these semantics."""
import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st

from gpu_docker_api_amd.state.mvcc import MemoryMVCC
from gpu_docker_api_amd.xerrors import NotExistInStore

KEYS = ["/a", "/b", "/p/x", "/p/y"]

ops = st.lists(
    st.one_of(
        st.tuples(st.just("put"), st.sampled_from(KEYS), st.text(max_size=4)),
        st.tuples(st.just("del"), st.sampled_from(KEYS), st.just("")),
    ),
    max_size=40,
)


@settings(max_examples=200, deadline=None)
@given(ops)
def test_get_at_rev_matches_snapshot_oracle(sequence):
    store = MemoryMVCC()
    snapshots = {1: {}}  # revision -> {key: value}
    state = {}
    for op, key, value in sequence:
        if op == "put":
            store.put(key, value)
            state[key] = value
            snapshots[store.revision] = dict(state)
        else:
            deleted = store.delete(key)
            if key in state:
                assert deleted == 1
                del state[key]
                snapshots[store.revision] = dict(state)
            else:
                assert deleted == 0

    # every historical revision must read back exactly the oracle snapshot
    for rev, snap in snapshots.items():
        for key in KEYS:
            if key in snap:
                assert store.get(key, rev=rev).value == snap[key]
            else:
                with pytest.raises(NotExistInStore):
                    store.get(key, rev=rev)


@settings(max_examples=200, deadline=None)
@given(ops)
def test_history_matches_put_log(sequence):
    store = MemoryMVCC()
    # oracle: per-key list of values of the current lifetime
    lifetimes = {k: [] for k in KEYS}
    for op, key, value in sequence:
        if op == "put":
            store.put(key, value)
            lifetimes[key].append(value)
        else:
            if store.delete(key):
                lifetimes[key] = []
    for key in KEYS:
        if lifetimes[key]:
            hist = store.history(key)
            assert [kv.value for kv in hist] == list(reversed(lifetimes[key]))
            assert [kv.version for kv in hist] == list(
                range(len(lifetimes[key]), 0, -1)
            )
        else:
            with pytest.raises(NotExistInStore):
                store.history(key)
