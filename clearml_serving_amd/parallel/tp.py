"""Tensor parallelism over RCCL/xGMI for LLM endpoints.

One process per GPU; ``torch.distributed`` backend "nccl" IS RCCL on ROCm.
Sharding (megatron-style, sized for xGMI):

- qkv / gate_up: column-parallel (head- and channel-sharded), no comm
- o_proj / down: row-parallel -> ONE all-reduce each per layer
- lm_head: column-parallel over vocab -> all-gather of logits shards

xGMI topology note: each MI355X has 7 point-to-point links
of ~153 GB/s; ring all-reduce is per-link bound. Decode activations are tiny
(batch x 4096 x 2 B), so RCCL's latency-oriented algorithms dominate; the
engine keeps ONE all-reduce per sublayer and batches sequences to amortize.
"""

from typing import Optional

import torch
import torch.distributed as dist


def is_initialized() -> bool:
    return dist.is_available() and dist.is_initialized()


def world_size() -> int:
    return dist.get_world_size() if is_initialized() else 1


def rank() -> int:
    return dist.get_rank() if is_initialized() else 0


def maybe_all_reduce(x: torch.Tensor, group=None) -> torch.Tensor:
    """Sum partial row-parallel outputs across the TP group (in place)."""
    if world_size() > 1:
        dist.all_reduce(x, group=group)
    return x


def gather_logits(logits: torch.Tensor, group=None) -> torch.Tensor:
    """All-gather column-sharded lm_head outputs -> full-vocab logits."""
    ws = world_size()
    if ws <= 1:
        return logits
    shards = [torch.empty_like(logits) for _ in range(ws)]
    dist.all_gather(shards, logits.contiguous(), group=group)
    return torch.cat(shards, dim=-1)


def broadcast_tokens(t: torch.Tensor, src: int = 0, group=None) -> torch.Tensor:
    if world_size() > 1:
        dist.broadcast(t, src=src, group=group)
    return t


def init_from_env(device_type: Optional[str] = None) -> int:
    """Initialize the process group from torchrun env (RANK/WORLD_SIZE);
    returns local rank. Safe to call when already initialized."""
    import os

    if not dist.is_available():
        return 0
    if not dist.is_initialized() and os.environ.get("WORLD_SIZE"):
        backend = "nccl" if (device_type or (
            "cuda" if torch.cuda.is_available() else "cpu")) == "cuda" \
            else "gloo"
        dist.init_process_group(backend=backend)
    local = int(os.environ.get("LOCAL_RANK", 0))
    if torch.cuda.is_available():
        torch.cuda.set_device(local)
    return local
