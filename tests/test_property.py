"""Property-based tests (hypothesis) for the routing/scheduling math."""

import math

from hypothesis import given, settings
from hypothesis import strategies as st


@settings(max_examples=200, deadline=None)
@given(weights=st.lists(st.floats(min_value=0.01, max_value=100.0),
                        min_size=1, max_size=8))
def test_canary_weight_normalization(weights):
    """Canary route weights always normalize to a probability vector."""
    from clearml_serving_amd.schemas import CanaryEP
    from clearml_serving_amd.serving.processor import ModelRequestProcessor
    from clearml_serving_amd.store import ServingStore

    import tempfile

    with tempfile.TemporaryDirectory() as d:
        store = ServingStore(d)
        p = ModelRequestProcessor(store=store, name="prop",
                                  force_create=True)
        eps = ["e/{}".format(i) for i in range(len(weights))]
        p.add_canary_endpoint(CanaryEP(endpoint="c", weights=list(weights),
                                       load_endpoints=eps))
        p._update_canary_lookup()
        route = p._canary_route["c"]
        total = sum(route["weights"])
        assert math.isclose(total, 1.0, rel_tol=1e-9)
        assert all(w >= 0 for w in route["weights"])
        assert len(route["endpoints"]) == len(weights)


@settings(max_examples=200, deadline=None)
@given(n=st.integers(min_value=1, max_value=64),
       buckets=st.lists(st.integers(min_value=1, max_value=64),
                        min_size=1, max_size=7))
def test_batcher_bucket_selection(n, buckets):
    """The chosen bucket is always >= the batch size and is the smallest
    qualifying preferred size."""
    from clearml_serving_amd.serving.batcher import DynamicBatcher

    b = DynamicBatcher(lambda x: x, device="cpu", max_batch_size=64,
                       preferred_batch_sizes=buckets, use_graphs=False)
    chosen = next(bk for bk in b.buckets if bk >= n)
    assert chosen >= n
    assert all(bk < n for bk in b.buckets if bk < chosen)
    assert b.buckets[-1] == 64  # max_batch_size always a bucket


@settings(max_examples=100, deadline=None)
@given(prompt=st.lists(st.integers(min_value=1, max_value=400),
                       min_size=1, max_size=60),
       block=st.sampled_from([4, 8, 16, 32]))
def test_block_allocator_slot_math(prompt, block):
    """Slot mapping is a bijection prompt-position -> (block, offset)."""
    from clearml_serving_amd.engines.llm.engine import BlockAllocator

    alloc = BlockAllocator(64)
    n_blocks = (len(prompt) + block - 1) // block
    blocks = alloc.alloc(n_blocks)
    slots = [blocks[p // block] * block + p % block
             for p in range(len(prompt))]
    assert len(set(slots)) == len(slots)
    for p, s in enumerate(slots):
        assert s // block == blocks[p // block]
    alloc.free(blocks)
    assert alloc.available == 64
