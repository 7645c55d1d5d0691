"""In-process HIP DL engine: the native replacement for the Triton sidecar.

Where the reference serializes numpy tensors into gRPC ``ModelInferRequest``
messages to a tritonserver container (reference: preprocess_service.py:
267-446) and configures batching via config.pbtxt pass-through
(triton_helper.py:291-409), this engine runs the model in the serving
process: ``process()`` awaits the per-endpoint dynamic batcher, which
coalesces concurrent requests into bucketed bf16 batches, executes on a
dedicated HIP stream (hipGraph-captured per bucket) and scatters outputs
back to the awaiting requests.

Model resolution, per endpoint:
1. user ``Preprocess.load()`` returned a callable -> use it directly
2. model file is a JSON model card -> built from the native model library
   (clearml_serving_amd.models: resnet50, bert-base, ...)
3. model file is TorchScript (.pt/.pth/.ts) -> torch.jit.load

``auxiliary_cfg`` knobs (the native equivalent of the reference's
``--aux-config`` pbtxt pass-through):
    max_batch_size (default 64), max_queue_delay_us (default 2000),
    preferred_batch_size (list), dtype ("bfloat16"|"float16"|"float32"),
    use_graphs (default true), gpu (device index), input_format ("dict")
"""

import os
from typing import Any, Dict, Optional

import numpy as np
import torch

from ..schemas import ModelEndpoint
from ..serving.batcher import DEFAULT_BUCKETS, DynamicBatcher
from ..serving.preprocess import BasePreprocessRequest

def _to_numpy(out):
    if isinstance(out, dict):
        return {k: _to_numpy(v) for k, v in out.items()}
    if isinstance(out, (tuple, list)):
        return [_to_numpy(v) for v in out]
    # bf16/fp16 need an fp32 hop; fp32/int tensors convert directly
    if out.dtype in (torch.bfloat16, torch.float16):
        out = out.float()
    return out.numpy()


_DTYPES = {
    "bfloat16": torch.bfloat16, "bf16": torch.bfloat16,
    "float16": torch.float16, "fp16": torch.float16,
    "float32": torch.float32, "fp32": torch.float32,
}


def _pick_device(aux: Dict[str, Any]) -> torch.device:
    """Endpoint -> GPU placement. Explicit via auxiliary_cfg {"gpu": N};
    otherwise the device with the most free HBM takes the model, so
    multi-model sessions spread across the node's 8 GPUs automatically
    (the reference delegates placement to Triton instance groups;
    SURVEY.md §2.6)."""
    if not torch.cuda.is_available():
        return torch.device("cpu")
    if "gpu" in aux:
        return torch.device("cuda", int(aux["gpu"]))
    best, best_free = 0, -1
    for i in range(torch.cuda.device_count()):
        free, _ = torch.cuda.mem_get_info(i)
        if free > best_free:
            best, best_free = i, free
    return torch.device("cuda", best)


@BasePreprocessRequest.register_engine("hip", modules=["torch"])
class HipPreprocessRequest(BasePreprocessRequest):
    is_preprocess_async = False
    is_process_async = True      # process() awaits the dynamic batcher
    is_postprocess_async = False

    def __init__(self, model_endpoint: ModelEndpoint, task=None):
        super().__init__(model_endpoint, task)
        aux = dict(model_endpoint.auxiliary_cfg or {})
        self.device = _pick_device(aux)
        if self.device.type == "cuda":
            # MIOpen find mode: per-shape conv algo search at first use
            # (batch shapes are bucketed, so the search cost is one-time at
            # warmup; measured 4.49 -> 2.90 ms on ResNet-50 b64 NHWC bf16)
            torch.backends.cudnn.benchmark = True
        self.dtype = _DTYPES.get(str(aux.get("dtype", "bfloat16")).lower(),
                                 torch.bfloat16)
        self._requested_dtype = self.dtype  # pre-CPU-override, for validation
        if self.device.type == "cpu" and self.dtype is not torch.float32:
            # CPU path (tests / no-GPU dev): fp32 keeps MIOpen-free numerics
            self.dtype = torch.float32

        model = self._model if callable(self._model) else None
        if model is None:
            model = self._load_model()
        self._torch_model = model

        buckets = aux.get("preferred_batch_size") or list(DEFAULT_BUCKETS)
        self._batcher = DynamicBatcher(
            model_fn=self._forward,
            device=self.device,
            max_batch_size=int(aux.get("max_batch_size", 64)),
            max_queue_delay_us=int(aux.get("max_queue_delay_us", 2000)),
            preferred_batch_sizes=buckets,
            use_graphs=bool(aux.get("use_graphs", True)),
            dtype=self.dtype,
            name=model_endpoint.serving_url,
            out_convert=_to_numpy,  # whole-batch bf16 -> float numpy, once
        )
        if aux.get("warmup", True):
            sample = self._sample_from_spec()
            if sample is not None:
                self._batcher.warmup(sample)

    def _sample_from_spec(self) -> Optional[Any]:
        """Build a zero request from the endpoint I/O spec so every batch
        bucket's hipGraph is captured before traffic arrives. Dynamic (-1)
        dims skip warmup (capture then happens lazily per shape)."""
        ep = self.model_endpoint
        if not ep.input_size or not ep.input_type:
            return None
        if any(d is None or int(d) < 0 for shape in ep.input_size
               for d in shape):
            return None
        tensors = []
        for shape, t in zip(ep.input_size, ep.input_type):
            np_dtype = np.dtype(t or "float32")
            tensors.append(torch.zeros([int(d) for d in shape],
                                       dtype=self._np_to_torch(np_dtype)))
        if ep.input_name and len(ep.input_name) == len(tensors):
            return dict(zip(ep.input_name, tensors))
        return tensors[0] if len(tensors) == 1 else None

    @staticmethod
    def _np_to_torch(np_dtype) -> torch.dtype:
        return {
            np.dtype("float32"): torch.float32,
            np.dtype("float16"): torch.float16,
            np.dtype("int64"): torch.int64,
            np.dtype("int32"): torch.int32,
            np.dtype("uint8"): torch.uint8,
        }.get(np_dtype, torch.float32)

    # ------------------------------------------------------------------ #
    def _load_model(self):
        local = self._get_local_model_file()
        if not local:
            raise ValueError(
                "hip endpoint '{}' has no model (register a model card or "
                "TorchScript file)".format(self.model_endpoint.serving_url))
        if os.path.isdir(local):
            for name in ("model_card.json", "card.json"):
                p = os.path.join(local, name)
                if os.path.exists(p):
                    local = p
                    break
        if local.endswith(".json"):
            if self._requested_dtype is torch.float16:
                # native model-library transformers run on the in-tree
                # attention kernels, which are bf16-only (attention.hip);
                # fail at endpoint construction with a clear message instead
                # of a TORCH_CHECK crash on the first GPU request
                raise ValueError(
                    "dtype=float16 is not supported for model-card endpoints "
                    "(the native attention kernels are bf16); set "
                    "dtype=bfloat16, or serve a TorchScript model for fp16")
            from .. import models

            model = models.build_model(local, device=str(self.device))
            if self.device.type == "cuda":
                model = model.to(self.dtype)
            else:
                model = model.float()
            return model
        model = torch.jit.load(local, map_location=str(self.device))
        model.eval()
        if self.device.type == "cuda":
            model = model.to(self.dtype)
        return model

    def _forward(self, batch):
        with torch.inference_mode():
            out = self._torch_model(batch)
        return out

    # ------------------------------------------------------------------ #
    @staticmethod
    def _to_tensor(value) -> torch.Tensor:
        if isinstance(value, torch.Tensor):
            return value
        arr = np.asarray(value)
        if arr.dtype == np.float64:
            arr = arr.astype(np.float32)
        return torch.from_numpy(arr)

    async def process(self, data: Any, state: dict,
                      collect_custom_statistics_fn=None) -> Any:
        """data: one request's input -- np array/list, or {name: array}."""
        if isinstance(data, dict):
            inputs = {k: self._to_tensor(v) for k, v in data.items()}
        else:
            inputs = self._to_tensor(data)
        # the batcher's out_convert already produced numpy (whole batch at
        # once); this request's slice is a view into that fresh array
        return await self._batcher.submit(inputs)


# alias for reference-CLI compatibility: `model add --engine triton` serves
# through the in-process HIP engine (there is no Triton sidecar to talk to)
BasePreprocessRequest.register_engine("triton")(HipPreprocessRequest)
BasePreprocessRequest.register_engine("pytorch")(HipPreprocessRequest)
