"""Native model library for the in-process HIP engine.

The reference serves DL models as opaque TorchScript/ONNX artifacts executed
by Triton. Here model families are first-class: architectures
are defined in-tree against the HIP op library (fused LayerNorm/attention/
epilogues), built from a JSON "model card" registered in the model registry:

    {"arch": "resnet50", "num_classes": 1000, "dtype": "bfloat16",
     "weights": "<optional safetensors path>"}

``build_model(card)`` returns an eval-mode torch.nn.Module ready for the
dynamic batcher (bf16 on GPU, hipGraph-capturable: no data-dependent python
control flow in forward).
"""

import json
import os
from typing import Optional, Union

import torch

_BUILDERS = {}


def register_arch(name: str):
    def deco(fn):
        _BUILDERS[name] = fn
        return fn

    return deco


def list_archs():
    return sorted(_BUILDERS.keys())


def build_model(card: Union[dict, str], device: Optional[str] = None) -> torch.nn.Module:
    """Build a model from a card dict or a path to a card JSON file."""
    if isinstance(card, str):
        with open(card, "rt") as f:
            card = json.load(f)
    card = dict(card)
    arch = card.pop("arch", None)
    if arch not in _BUILDERS:
        raise ValueError(
            "unknown arch '{}' (known: {})".format(arch, list_archs()))
    weights = card.pop("weights", None)
    dtype = getattr(torch, card.pop("dtype", "bfloat16"))
    model = _BUILDERS[arch](**card)
    if weights:
        load_weights(model, weights)
    model = model.eval()
    if device is not None:
        model = model.to(device)
    if dtype in (torch.bfloat16, torch.float16):
        model = model.to(dtype)
    # post-dtype fusion hooks (e.g. ResNet BN folding happens pre-dtype)
    return model


def load_weights(model: torch.nn.Module, path: str) -> None:
    if path.endswith(".safetensors"):
        from safetensors.torch import load_file

        state = load_file(path)
    else:
        state = torch.load(path, map_location="cpu", weights_only=True)
    # raw HuggingFace checkpoints convert on the fly (key-layout
    # detection; native layouts pass through) -- a hub download drops in
    # without a manual conversion step
    from .convert import convert_hf_auto

    model.load_state_dict(convert_hf_auto(state))


from . import bert, gpt2, llama, resnet  # noqa: E402,F401  (register architectures)
