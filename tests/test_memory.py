import numpy as np
import pytest

from trtlab_amd.memory import ArenaPlanner, TransactionalStack


def test_transactional_stack():
    st = TransactionalStack(10240, alignment=256)
    a = st.allocate(100)
    assert a == 0
    st.begin()
    b = st.allocate(100)
    assert b == 256
    st.commit()
    c = st.allocate(100)
    assert c == 256  # reused after transaction rollback
    with pytest.raises(MemoryError):
        st.allocate(1 << 20)


def _check_no_overlap(items, offsets):
    # items: list of (name, size, start, end)
    for i, (n1, s1, a1, b1) in enumerate(items):
        for n2, s2, a2, b2 in items[i + 1:]:
            if b1 < a2 or b2 < a1:
                continue  # lifetimes disjoint
            o1, o2 = offsets[n1], offsets[n2]
            assert o1 + s1 <= o2 or o2 + s2 <= o1, f"overlap {n1} {n2}"


def test_arena_planner_no_overlap():
    rng = np.random.RandomState(0)
    items = []
    planner = ArenaPlanner(alignment=256)
    for i in range(200):
        size = int(rng.randint(1, 1 << 16))
        start = int(rng.randint(0, 50))
        end = start + int(rng.randint(0, 10))
        name = f"t{i}"
        items.append((name, ((size + 255) // 256) * 256, start, end))
        planner.add(name, size, start, end)
    offsets, total = planner.plan()
    assert total % 256 == 0
    _check_no_overlap(items, offsets)
    # reuse should beat the no-reuse sum
    assert total < sum(s for _, s, _, _ in items)


def test_arena_planner_serial_chain_reuses():
    planner = ArenaPlanner()
    # a -> b -> c : a dies when b is made, so c can reuse a's space
    planner.add("a", 1000, 0, 1)
    planner.add("b", 1000, 1, 2)
    planner.add("c", 1000, 2, 3)
    offsets, total = planner.plan()
    assert total <= 2 * 1024


def test_huge_page_buffer_host():
    """Huge-page host allocator: 2 MiB rounding, zero-copy views, byte
    accounting (reference trtlab/memory huge-page raw allocator)."""
    from trtlab_amd import native
    from trtlab_amd.memory import HugePageBuffer

    C = native()
    before = C.memory.huge_bytes_in_use()
    b = HugePageBuffer(3 << 20)  # rounds to 2 x 2 MiB
    assert C.memory.huge_bytes_in_use() - before == 4 << 20
    assert isinstance(b.hugetlb, bool)  # explicit hugetlb or THP fallback
    a = b.array(np.float32)
    a[:] = 1.5
    assert float(b.array(np.float32)[-1]) == 1.5
    b.close()
    assert C.memory.huge_bytes_in_use() == before


@pytest.mark.gpu
def test_huge_page_buffer_pinned_gpu():
    """pin=True hipHostRegisters the range: usable for GPU DMA."""
    from trtlab_amd.memory import DeviceBuffer, HugePageBuffer

    b = HugePageBuffer(1 << 20, pin=True)
    a = b.array(np.float32)
    a[:] = 2.25
    d = DeviceBuffer(1 << 20)
    d.upload(a)
    out = np.zeros(1 << 18, np.float32)
    d.download(out)
    assert out[0] == 2.25 and out[-1] == 2.25
    d.close()
    b.close()


def test_arena_planner_reuses_disjoint_lifetimes():
    """Best-fit arena: tensors with disjoint lifetimes share bytes, total
    stays below the no-reuse sum (the property the engine's activation
    memory depends on)."""
    ap = ArenaPlanner()
    ap.add("a", 1000, 0, 2)
    ap.add("b", 1000, 3, 5)   # disjoint from a -> can share
    ap.add("c", 1000, 1, 4)   # overlaps both
    offsets, total = ap.plan()
    assert offsets["a"] == offsets["b"]
    assert offsets["c"] != offsets["a"]
    assert total < 3 * 1024  # reuse happened (256-B aligned blocks)


def test_transactional_stack_nesting():
    st = TransactionalStack(4096)
    st.begin()
    st.allocate(100)
    st.begin()
    st.allocate(100)
    st.commit()   # inner rollback point
    inner_top = st.high_water
    st.commit()
    assert st.high_water < inner_top


class _FakeInner:
    """CPU stand-in for a device allocator (composability tests)."""

    def __init__(self):
        self.n = 0
        self.live = set()

    def allocate(self, nbytes):
        self.n += 1
        self.live.add(self.n)
        return self.n

    def deallocate(self, ptr):
        self.live.discard(ptr)


def test_count_limited_allocator():
    from trtlab_amd.memory import CountLimitedAllocator

    a = CountLimitedAllocator(_FakeInner(), max_count=2)
    p1, p2 = a.allocate(10), a.allocate(10)
    with pytest.raises(MemoryError):
        a.allocate(10)
    a.deallocate(p1)
    a.allocate(10)  # slot freed


def test_size_limited_allocator():
    from trtlab_amd.memory import SizeLimitedAllocator

    a = SizeLimitedAllocator(_FakeInner(), max_bytes=100)
    p = a.allocate(80)
    with pytest.raises(MemoryError):
        a.allocate(30)
    a.deallocate(p)
    a.allocate(95)
    assert a.in_use == 95


def test_tracked_allocator_composes():
    from trtlab_amd.memory import (CountLimitedAllocator, TrackedAllocator)

    a = TrackedAllocator(CountLimitedAllocator(_FakeInner(), 8), "t")
    ps = [a.allocate(1 << i) for i in range(4)]
    for p in ps:
        a.deallocate(p)
    assert a.tracker.total_allocs == 4
    assert a.tracker.in_use == 0
    assert a.tracker.high_water == (1 + 2 + 4 + 8)


def test_hybrid_futex_mutex_stress():
    """Hybrid spin-then-futex mutex + condvar (reference hybrid_mutex.h
    role): 8 threads x 200k guarded increments must lose no updates, and
    the condvar start-gate must release every thread."""
    import trtlab_amd

    C = trtlab_amd.native()
    n, ms = C.memory.hybrid_mutex_stress(8, 200000)
    assert n == 8 * 200000
    assert ms < 60000


def test_first_touch_buffer():
    """NUMA first-touch: pages fault on a thread pinned to the target
    node's CPUs (reference first_touch_allocator.h:35). On single-node
    or affinity-restricted machines the pin degrades gracefully."""
    from trtlab_amd.core.numa import NumaTopology
    from trtlab_amd.memory import FirstTouchBuffer

    b = FirstTouchBuffer(1 << 20)  # plain first-touch, calling thread
    assert b.ptr != 0 and b.nbytes == 1 << 20
    import ctypes
    ctypes.memset(b.ptr, 0xAB, 64)  # pages are mapped and writable
    assert ctypes.string_at(b.ptr, 2) == b"\xab\xab"
    b.close()

    topo = NumaTopology()
    nodes = topo.nodes if hasattr(topo, "nodes") else []
    node0 = (nodes[0].id if nodes and hasattr(nodes[0], "id")
             else (0 if nodes else 0))
    b2 = FirstTouchBuffer(1 << 16, node=node0, topology=topo)
    assert b2.ptr != 0
    if b2.touched_on:  # affinity honored: touch thread ran on node CPUs
        allowed = set(topo.nearest_cpus(node0))
        assert set(b2.touched_on) <= allowed or not allowed
    b2.close()
