"""MVCC store semantics — must match etcd exactly."""
import os

import pytest

from gpu_docker_api_amd.state.mvcc import MemoryMVCC
from gpu_docker_api_amd.state.wal import Wal
from gpu_docker_api_amd.xerrors import NotExistInStore, RevisionCompacted


def test_first_put_lands_at_revision_2():
    s = MemoryMVCC()
    kv = s.put("/a", "1")
    assert s.revision == 2
    assert kv.create_revision == 2
    assert kv.mod_revision == 2
    assert kv.version == 1


def test_version_and_mod_revision_advance_per_put():
    s = MemoryMVCC()
    s.put("/a", "1")
    s.put("/b", "x")
    kv = s.put("/a", "2")
    assert kv.create_revision == 2
    assert kv.mod_revision == 4
    assert kv.version == 2


def test_get_at_revision_sees_past_state():
    s = MemoryMVCC()
    s.put("/a", "1")   # rev 2
    s.put("/a", "2")   # rev 3
    s.put("/a", "3")   # rev 4
    assert s.get("/a", rev=2).value == "1"
    assert s.get("/a", rev=3).value == "2"
    assert s.get("/a").value == "3"
    # between revisions: state as of that revision
    s.put("/other", "x")  # rev 5
    assert s.get("/a", rev=5).value == "3"


def test_delete_leaves_tombstone_and_resets_lifetime():
    s = MemoryMVCC()
    s.put("/a", "1")       # rev 2, v1
    s.put("/a", "2")       # rev 3, v2
    assert s.delete("/a") == 1   # rev 4
    with pytest.raises(NotExistInStore):
        s.get("/a")
    # history before the delete is still readable at-revision
    assert s.get("/a", rev=3).value == "2"
    # recreate: version restarts at 1, create_revision is new
    kv = s.put("/a", "3")  # rev 5
    assert kv.version == 1
    assert kv.create_revision == 5
    assert s.delete("/a") == 1
    assert s.delete("/a") == 0  # double delete is a no-op


def test_history_newest_first_current_lifetime_only():
    s = MemoryMVCC()
    s.put("/k", "old1")
    s.delete("/k")
    s.put("/k", "a")  # v1
    s.put("/k", "b")  # v2
    s.put("/k", "c")  # v3
    hist = s.history("/k")
    assert [kv.value for kv in hist] == ["c", "b", "a"]
    assert [kv.version for kv in hist] == [3, 2, 1]
    assert s.get_version("/k", 2).value == "b"
    with pytest.raises(NotExistInStore):
        s.get_version("/k", 9)


def test_history_of_missing_or_deleted_key_raises():
    s = MemoryMVCC()
    with pytest.raises(NotExistInStore):
        s.history("/nope")
    s.put("/k", "a")
    s.delete("/k")
    with pytest.raises(NotExistInStore):
        s.history("/k")


def test_range_prefix_sorted_and_at_revision():
    s = MemoryMVCC()
    s.put("/p/b", "1")  # rev 2
    s.put("/p/a", "2")  # rev 3
    s.put("/q/z", "3")  # rev 4
    s.put("/p/a", "4")  # rev 5
    kvs = s.range_prefix("/p/")
    assert [(kv.key, kv.value) for kv in kvs] == [("/p/a", "4"), ("/p/b", "1")]
    kvs_old = s.range_prefix("/p/", rev=3)
    assert [(kv.key, kv.value) for kv in kvs_old] == [("/p/a", "2"), ("/p/b", "1")]


def test_delete_prefix_single_transaction():
    s = MemoryMVCC()
    s.put("/p/a", "1")
    s.put("/p/b", "2")
    s.put("/q/c", "3")
    rev_before = s.revision
    assert s.delete_prefix("/p/") == 2
    assert s.revision == rev_before + 1  # one txn for the whole range
    assert s.range_prefix("/p/") == []
    assert s.get("/q/c").value == "3"


def test_compaction_blocks_old_reads_keeps_floor():
    s = MemoryMVCC()
    s.put("/a", "1")  # rev 2
    s.put("/a", "2")  # rev 3
    s.put("/a", "3")  # rev 4
    s.compact(3)
    # reading at/after the compaction revision still works
    assert s.get("/a", rev=3).value == "2"
    assert s.get("/a", rev=4).value == "3"
    with pytest.raises(RevisionCompacted):
        s.get("/a", rev=2)
    # history hides events below the compaction floor but keeps the floor
    hist = s.history("/a")
    assert [kv.value for kv in hist] == ["3", "2"]


def test_future_revision_read_raises():
    s = MemoryMVCC()
    s.put("/a", "1")
    with pytest.raises(NotExistInStore):
        s.get("/a", rev=99)


def test_wal_replay_restores_full_history(tmp_path):
    path = str(tmp_path / "wal.jsonl")
    s = MemoryMVCC()
    Wal(path).attach(s)
    s.put("/a", "1")
    s.put("/a", "2")
    s.put("/b", "x")
    s.delete("/b")
    rev = s.revision

    s2 = MemoryMVCC()
    Wal(path).attach(s2)
    assert s2.revision == rev
    assert s2.get("/a").value == "2"
    assert [kv.value for kv in s2.history("/a")] == ["2", "1"]
    with pytest.raises(NotExistInStore):
        s2.get("/b")
    # and the reopened store keeps appending correctly
    s2.put("/a", "3")
    assert s2.get("/a").version == 3


def test_wal_torn_tail_is_ignored(tmp_path):
    path = str(tmp_path / "wal.jsonl")
    s = MemoryMVCC()
    w = Wal(path)
    w.attach(s)
    s.put("/a", "1")
    w.close()
    with open(path, "a") as f:
        f.write('{"r": 3, "k": "/a", "v": "tor')  # torn crash write
    s2 = MemoryMVCC()
    Wal(path).attach(s2)
    assert s2.get("/a").value == "1"
