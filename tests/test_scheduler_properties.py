"""Property-based scheduler invariants: random apply/restore sequences can
never double-allocate, leak, or corrupt counts."""
import asyncio

import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st

from gpu_docker_api_amd.parallel import CpuScheduler, GpuScheduler, MockInventory, PortScheduler
from gpu_docker_api_amd.state import MemoryStore
from gpu_docker_api_amd.xerrors import CpuNotEnough, GpuNotEnough, PortNotEnough

ops = st.lists(
    st.one_of(
        st.tuples(st.just("apply"), st.integers(min_value=0, max_value=10)),
        st.tuples(st.just("restore"), st.integers(min_value=0, max_value=5)),
    ),
    max_size=50,
)


@settings(max_examples=100, deadline=None)
@given(ops)
def test_gpu_scheduler_never_double_allocates(sequence):
    async def main():
        gs = await GpuScheduler.create(MemoryStore(), None, MockInventory(8))
        held: list = []  # list of allocations (each a list of uuids)
        for op, n in sequence:
            if op == "apply":
                free = 8 - sum(gs.get_gpu_status().values())
                if n <= 0 or n > 8:
                    with pytest.raises(GpuNotEnough):
                        gs.apply(n)  # apply(<=0) raises too
                elif n > free:
                    with pytest.raises(GpuNotEnough):
                        gs.apply(n)
                else:
                    got = gs.apply(n)
                    assert len(got) == n
                    # none of them was already held
                    flat = [u for a in held for u in a]
                    assert not (set(got) & set(flat))
                    held.append(got)
            else:  # restore the n-th oldest allocation if it exists
                if n < len(held):
                    gs.restore(held.pop(n))
            # invariant: used count == held count
            assert sum(gs.get_gpu_status().values()) == sum(len(a) for a in held)

    asyncio.run(main())


@settings(max_examples=100, deadline=None)
@given(ops)
def test_cpu_scheduler_invariants(sequence):
    async def main():
        cs = await CpuScheduler.create(MemoryStore(), None, count=8)
        held: list = []
        for op, n in sequence:
            if op == "apply":
                free = 8 - sum(cs.get_cpu_status().values())
                if n <= 0 or n > 8 or n > free:
                    if n != 0:
                        with pytest.raises(CpuNotEnough):
                            cs.apply(n)
                else:
                    cpuset = cs.apply(n)
                    ids = cpuset.split(",")
                    assert len(ids) == n == len(set(ids))
                    flat = [i for a in held for i in a]
                    assert not (set(ids) & set(flat))
                    held.append(ids)
            else:
                if n < len(held):
                    cs.restore(held.pop(n))
            assert sum(cs.get_cpu_status().values()) == sum(len(a) for a in held)

    asyncio.run(main())


@settings(max_examples=60, deadline=None)
@given(ops)
def test_port_scheduler_invariants(sequence):
    async def main():
        ps = await PortScheduler.create(MemoryStore(), None, 40000, 40015)  # 16 ports
        held: list = []
        for op, n in sequence:
            if op == "apply":
                free = ps.get_port_status()["AvailableCount"]
                if n <= 0 or n > 16 or n > free:
                    if n != 0:
                        with pytest.raises(PortNotEnough):
                            ps.apply(n)
                else:
                    got = ps.apply(n)
                    assert len(got) == n == len(set(got))
                    flat = [p for a in held for p in a]
                    assert not (set(got) & set(flat))
                    held.append(got)
            else:
                if n < len(held):
                    ps.restore(held.pop(n))
            assert ps.get_port_status()["AvailableCount"] == 16 - sum(len(a) for a in held)

    asyncio.run(main())
