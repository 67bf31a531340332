"""Property test of the refcounted block allocator — the invariant the
whole paged-KV design rests on (shared prompt blocks across the
n-candidate fan-out)."""

from hypothesis import given, settings
from hypothesis import strategies as st

from distrl_llm_amd.engine.kvcache import BlockAllocator


@settings(max_examples=60, deadline=None)
@given(st.lists(st.tuples(st.sampled_from(["alloc", "incref", "free"]),
                          st.integers(min_value=0, max_value=10 ** 6)),
                max_size=200),
       st.integers(min_value=1, max_value=32))
def test_allocator_refcount_invariants(ops, num_blocks):
    """Random alloc/incref/free interleavings: refcounts never negative,
    free list and refcounts always consistent, no double-free of a block
    back into the free list, full drain restores every block."""
    a = BlockAllocator(num_blocks)
    live = {}  # block -> expected refcount

    for op, arg in ops:
        if op == "alloc":
            if a.num_free == 0:
                continue
            b = a.alloc()
            assert b not in live
            live[b] = 1
        elif live:
            b = sorted(live)[arg % len(live)]
            if op == "incref":
                a.incref(b)
                live[b] += 1
            else:
                a.free(b)
                live[b] -= 1
                if live[b] == 0:
                    del live[b]
        # invariants after every step
        assert a.num_free == num_blocks - len(live)
        for b, rc in live.items():
            assert a.refcount(b) == rc

    for b in list(live):
        for _ in range(live[b]):
            a.free(b)
    assert a.num_free == num_blocks

    a2 = BlockAllocator(num_blocks)
    a2.reset()
    assert a2.num_free == num_blocks
