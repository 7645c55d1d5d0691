"""Native model library + hip engine + dynamic batcher (CPU execution path)."""

import asyncio
import json

import numpy as np
import pytest
import torch

from clearml_serving_amd.models import build_model, list_archs
from clearml_serving_amd.schemas import ModelEndpoint
from clearml_serving_amd.serving.batcher import DynamicBatcher


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        # cancel lingering workers (batcher/engine loops) before closing
        pending = asyncio.all_tasks(loop)
        for t in pending:
            t.cancel()
        if pending:
            loop.run_until_complete(
                asyncio.gather(*pending, return_exceptions=True))
        loop.close()


def test_arch_registry():
    for a in ("resnet50", "bert-base"):
        assert a in list_archs()


def test_resnet50_builds_and_runs():
    m = build_model({"arch": "resnet50", "num_classes": 10, "dtype": "float32"})
    x = torch.randn(2, 3, 64, 64)
    with torch.inference_mode():
        y = m(x)
    assert y.shape == (2, 10)
    assert torch.isfinite(y).all()


def test_bert_base_builds_and_runs():
    m = build_model({"arch": "bert-base", "num_labels": 3, "dtype": "float32",
                     "vocab_size": 1000})
    ids = torch.randint(0, 1000, (2, 16))
    mask = torch.ones(2, 16, dtype=torch.int32)
    with torch.inference_mode():
        y = m({"input_ids": ids, "attention_mask": mask})
    assert y.shape == (2, 3)
    assert torch.isfinite(y).all()


def test_bert_padding_mask_matters():
    torch.manual_seed(0)
    m = build_model({"arch": "bert-base", "num_labels": 2, "dtype": "float32",
                     "vocab_size": 100})
    ids = torch.randint(0, 100, (1, 16))
    full = torch.ones(1, 16, dtype=torch.int32)
    half = torch.cat([torch.ones(1, 8, dtype=torch.int32),
                      torch.zeros(1, 8, dtype=torch.int32)], dim=1)
    with torch.inference_mode():
        y_full = m({"input_ids": ids, "attention_mask": full})
        y_half = m({"input_ids": ids, "attention_mask": half})
        # padding-masked forward must equal the truncated forward
        y_trunc = m({"input_ids": ids[:, :8],
                     "attention_mask": torch.ones(1, 8, dtype=torch.int32)})
    assert not torch.allclose(y_full, y_half, atol=1e-4)
    torch.testing.assert_close(y_half, y_trunc, atol=1e-4, rtol=1e-4)


# ------------------------------------------------------------------ #
# dynamic batcher
# ------------------------------------------------------------------ #
def test_batcher_coalesces_and_slices():
    calls = []

    def model_fn(x):
        calls.append(x.shape[0])
        return x * 2.0

    batcher = DynamicBatcher(model_fn, device="cpu", max_batch_size=8,
                             max_queue_delay_us=50_000, use_graphs=False)

    async def main():
        outs = await asyncio.gather(
            *[batcher.submit(torch.full((3,), float(i))) for i in range(6)])
        return outs

    outs = run(main())
    for i, o in enumerate(outs):
        torch.testing.assert_close(o, torch.full((3,), 2.0 * i))
    # everything coalesced into few batches, padded to a bucket
    assert sum(calls) >= 6
    assert len(calls) <= 3


def test_batcher_bucket_padding():
    shapes = []

    def model_fn(x):
        shapes.append(tuple(x.shape))
        return x

    batcher = DynamicBatcher(model_fn, device="cpu", max_batch_size=16,
                             max_queue_delay_us=30_000, use_graphs=False)

    async def main():
        return await asyncio.gather(
            *[batcher.submit(torch.zeros(2)) for _ in range(3)])

    run(main())
    # 3 requests pad up to bucket size 4
    assert all(s[0] in (1, 2, 4) for s in shapes)


def test_batcher_dict_inputs():
    def model_fn(d):
        return d["a"] + d["b"]

    batcher = DynamicBatcher(model_fn, device="cpu", max_batch_size=4,
                             max_queue_delay_us=10_000, use_graphs=False)

    async def main():
        return await batcher.submit(
            {"a": torch.ones(2), "b": torch.full((2,), 3.0)})

    out = run(main())
    torch.testing.assert_close(out, torch.full((2,), 4.0))


def test_batcher_error_propagates():
    def model_fn(x):
        raise RuntimeError("boom")

    batcher = DynamicBatcher(model_fn, device="cpu", use_graphs=False,
                             max_queue_delay_us=1000)

    async def main():
        with pytest.raises(RuntimeError, match="boom"):
            await batcher.submit(torch.zeros(1))

    run(main())


# ------------------------------------------------------------------ #
# hip engine through the full serving path (CPU device)
# ------------------------------------------------------------------ #
@pytest.fixture()
def model_card_registered(store, tmp_path):
    card = tmp_path / "model_card.json"
    card.write_text(json.dumps({
        "arch": "bert-base", "num_labels": 2, "dtype": "float32",
        "vocab_size": 500,
    }))
    return store.register_model(name="bert-tiny-card", project="p",
                                path=str(card))


def test_hip_engine_serves_model_card(processor, store, model_card_registered):
    processor.add_endpoint(ModelEndpoint(
        engine_type="hip", serving_url="bert_ep",
        model_id=model_card_registered.model_id,
        auxiliary_cfg={"max_queue_delay_us": 1000, "use_graphs": False},
    ))
    body = {
        "input_ids": np.random.randint(0, 500, (16,)).tolist(),
        "attention_mask": [1] * 16,
    }
    out = run(processor.process_request("bert_ep", "", body))
    assert np.asarray(out).shape == (2,)


def test_triton_alias_serves_same_engine(processor, store, model_card_registered):
    processor.add_endpoint(ModelEndpoint(
        engine_type="triton", serving_url="bert_ep2",
        model_id=model_card_registered.model_id,
        auxiliary_cfg={"max_queue_delay_us": 1000, "use_graphs": False},
    ))
    out = run(processor.process_request("bert_ep2", "", {
        "input_ids": [1, 2, 3, 4], "attention_mask": [1, 1, 1, 1]}))
    assert np.asarray(out).shape == (2,)


def test_torchscript_model_serving(processor, store, tmp_path):
    class TinyNet(torch.nn.Module):
        def forward(self, x):
            return x.sum(dim=-1, keepdim=True) * 2.0

    path = tmp_path / "tiny.pt"
    torch.jit.script(TinyNet()).save(str(path))
    rec = store.register_model(name="ts-model", project="p", path=str(path))
    processor.add_endpoint(ModelEndpoint(
        engine_type="hip", serving_url="ts_ep", model_id=rec.model_id,
        auxiliary_cfg={"max_queue_delay_us": 1000, "use_graphs": False,
                       "dtype": "float32"},
    ))
    out = run(processor.process_request("ts_ep", "", [1.0, 2.0, 3.0]))
    assert abs(float(np.asarray(out).ravel()[0]) - 12.0) < 1e-5


def test_hip_engine_auto_update_version_swap(processor, store, tmp_path):
    """Auto-update materializes hip-engine versions; newest serves after a
    registry update (the Triton-sidecar repo-sync equivalent)."""
    from clearml_serving_amd.schemas import ModelMonitoring

    def card(labels):
        p = tmp_path / ("c%d.json" % labels)
        p.write_text(json.dumps({"arch": "bert-base", "num_labels": labels,
                                 "dtype": "float32", "vocab_size": 100}))
        return p

    processor.add_model_monitoring(ModelMonitoring(
        base_serving_url="hip_auto", engine_type="hip",
        monitor_name="^bert-auto$", max_versions=1,
        auxiliary_cfg={"max_queue_delay_us": 500, "use_graphs": False}))
    store.register_model(name="bert-auto", project="p", path=str(card(2)))
    processor._update_monitored_models()
    body = {"input_ids": [1, 2, 3], "attention_mask": [1, 1, 1]}
    out = run(processor.process_request("hip_auto", "1", body))
    assert np.asarray(out).shape == (2,)

    import time as _t

    _t.sleep(0.01)
    store.register_model(name="bert-auto", project="p", path=str(card(3)))
    processor._update_monitored_models()
    assert "hip_auto/2" in processor.get_synced_endpoints()
    assert "hip_auto/1" not in processor.get_synced_endpoints()
    out = run(processor.process_request("hip_auto", "2", body))
    assert np.asarray(out).shape == (3,)


def test_batcher_respects_queue_delay():
    """A lone request waits at most ~max_queue_delay before executing."""
    import time as _t

    def model_fn(x):
        return x

    batcher = DynamicBatcher(model_fn, device="cpu", max_batch_size=64,
                             max_queue_delay_us=30_000, use_graphs=False)

    async def main():
        t0 = _t.monotonic()
        await batcher.submit(torch.zeros(2))
        return _t.monotonic() - t0

    elapsed = run(main())
    # generous upper bound: loaded CI boxes schedule slowly; the point
    # is ONLY that the lone request didn't wait forever
    assert 0.02 <= elapsed < 3.0


def test_batcher_worker_restarts_after_crash():
    """A worker killed by an unexpected exception is replaced on the next
    submit instead of hanging every future."""
    def model_fn(x):
        return x * 2

    batcher = DynamicBatcher(model_fn, device="cpu", max_batch_size=4,
                             max_queue_delay_us=1000, use_graphs=False)

    async def main():
        out = await batcher.submit(torch.ones(2))
        # kill the worker the way a code bug would
        batcher._worker_task.cancel()
        try:
            await batcher._worker_task
        except asyncio.CancelledError:
            pass
        out2 = await batcher.submit(torch.ones(2))
        return out, out2

    out, out2 = run(main())
    assert torch.equal(out, out2)


def test_hip_engine_honors_triton_style_batch_config(processor, store,
                                                     tmp_path):
    """The reference passes Triton dynamic_batching via aux config
    (preferred_batch_size, max_batch_size, max_queue_delay -- huggingface
    example readme:113); the hip engine maps the same keys onto the
    DynamicBatcher."""
    import json

    card = tmp_path / "card.json"
    card.write_text(json.dumps({"arch": "resnet50"}))
    rec = store.register_model(name="r50", project="p", path=str(card))
    processor.add_endpoint(ModelEndpoint(
        engine_type="hip", serving_url="batchcfg", model_id=rec.model_id,
        input_size=[3, 32, 32], input_type="float32",
        auxiliary_cfg={"preferred_batch_size": [1, 2, 4, 8],
                       "max_batch_size": 8, "max_queue_delay_us": 1234,
                       "use_graphs": False, "device": "cpu"}))
    out = run(processor.process_request(
        "batchcfg", "", np.zeros((3, 32, 32), dtype=np.float32)))
    assert out is not None
    engine = processor._engine_processor_lookup["batchcfg"]
    b = engine._batcher
    assert b.max_batch_size == 8
    assert b.buckets == [1, 2, 4, 8]
    assert abs(b.max_queue_delay_s - 0.001234) < 1e-9


def test_pick_device_most_free_hbm(monkeypatch):
    """Free-HBM placement: multi-model sessions spread across GPUs."""
    from clearml_serving_amd.engines import torch_engine as te

    monkeypatch.setattr(torch.cuda, "is_available", lambda: True)
    monkeypatch.setattr(torch.cuda, "device_count", lambda: 4)
    free = {0: 10, 1: 40, 2: 25, 3: 40}
    monkeypatch.setattr(torch.cuda, "mem_get_info",
                        lambda i: (free[i], 100))
    assert te._pick_device({}) == torch.device("cuda", 1)  # first max wins
    assert te._pick_device({"gpu": 3}) == torch.device("cuda", 3)


def test_fp16_model_card_endpoint_rejected(processor, store,
                                           model_card_registered):
    """dtype=float16 on a model-card endpoint fails at construction with a
    clear error (native attention kernels are bf16-only) instead of a
    TORCH_CHECK crash on the first GPU request."""
    processor.add_endpoint(ModelEndpoint(
        engine_type="hip", serving_url="bert_fp16",
        model_id=model_card_registered.model_id,
        auxiliary_cfg={"dtype": "float16", "use_graphs": False},
    ))
    body = {"input_ids": [1, 2, 3], "attention_mask": [1] * 3}
    with pytest.raises(ValueError, match="float16"):
        run(processor.process_request("bert_fp16", "", body))


def test_fp16_torchscript_endpoint_allowed(processor, store, tmp_path):
    """TorchScript fp16 endpoints stay supported (torch-native ops handle
    fp16; only the in-tree attention path is bf16-only)."""

    class M(torch.nn.Module):
        def forward(self, x):
            return x * 2.0

    p = tmp_path / "m.pt"
    torch.jit.script(M()).save(str(p))
    rec = store.register_model(name="dbl", project="p", path=str(p))
    processor.add_endpoint(ModelEndpoint(
        engine_type="hip", serving_url="dbl_ep", model_id=rec.model_id,
        auxiliary_cfg={"dtype": "float16", "use_graphs": False,
                       "max_queue_delay_us": 500},
    ))
    out = run(processor.process_request("dbl_ep", "", [1.0, 2.0]))
    assert np.allclose(np.asarray(out), [2.0, 4.0])


def test_batcher_heterogeneous_requests_do_not_poison_neighbors():
    """A coalescing window can catch requests with different shapes or
    key sets; each signature group dispatches separately, so every
    request gets its own correct result instead of one stack failure
    500ing the whole window."""
    import asyncio

    import torch

    from clearml_serving_amd.serving.batcher import DynamicBatcher

    def model(x):
        if isinstance(x, dict):
            return x["a"] * 2
        return x + 1

    b = DynamicBatcher(model_fn=model, device="cpu", max_batch_size=16,
                       max_queue_delay_us=30000, use_graphs=False)

    async def main():
        reqs = [
            torch.ones(3),                 # shape (3,)
            torch.ones(5),                 # shape (5,) -- different!
            {"a": torch.ones(2)},          # dict input
            torch.ones(3) * 2,
            {"a": torch.ones(2) * 3},
        ]
        return await asyncio.gather(*[b.submit(r) for r in reqs])

    loop = asyncio.new_event_loop()
    try:
        outs = loop.run_until_complete(main())
    finally:
        for t in asyncio.all_tasks(loop):
            t.cancel()
        loop.run_until_complete(asyncio.sleep(0))
        loop.close()
    torch.testing.assert_close(outs[0], torch.ones(3) + 1)
    torch.testing.assert_close(outs[1], torch.ones(5) + 1)
    torch.testing.assert_close(outs[2], torch.ones(2) * 2)
    torch.testing.assert_close(outs[3], torch.ones(3) * 2 + 1)
    torch.testing.assert_close(outs[4], torch.ones(2) * 6)
