"""Fixed-layout step-plan broadcast for tensor-parallel serving.

Rank 0 schedules; worker ranks follow. Round 1 broadcast the per-step plan
dict with ``broadcast_object_list`` (pickle + CPU sync) -- fatal inside a
few-ms TP=8 decode step budget. This codec packs every plan mode into ONE
preallocated int32 tensor and broadcasts it with a single collective
(RCCL on GPU, gloo on CPU); floats travel as bit-cast int32.

Layout (int32 words):
  [0] mode  (0 stop | 1 prefill | 2 chunk | 3 decode | 4 embed | 5 object)
  [1] b     (rows)
  [2..7]    mode-specific scalars
  [8..]     mode-specific payload (see _encode_* below)

Anything that exceeds the fixed buffer (giant embed batches) falls back to
mode=object + ``broadcast_object_list`` -- off the steady-state path.
"""

from typing import Any, Dict, Optional

import numpy as np
import torch

MODE_STOP = 0
MODE_PREFILL = 1
MODE_CHUNK = 2
MODE_DECODE = 3
MODE_EMBED = 4
MODE_OBJECT = 5

_HDR = 8


def _f2i(vals) -> np.ndarray:
    return np.asarray(vals, dtype=np.float32).view(np.int32)


def _i2f(arr: np.ndarray) -> np.ndarray:
    return np.asarray(arr, dtype=np.int32).view(np.float32)


class PlanCodec:
    def __init__(self, cfg, device: torch.device):
        self.device = device
        S = int(cfg.max_num_seqs)
        MB = (int(cfg.max_model_len) + int(cfg.block_size) - 1) \
            // int(cfg.block_size)
        P = max(int(cfg.max_prefill_tokens), int(cfg.prefill_chunk))
        # worst cases: decode = 12S + S*MB; prefill = 8S + 2P;
        # chunk = 13S + 2P + S*MB. Embed batches that don't fit fall back
        # to the object path (they are OFF the per-token hot loop) -- the
        # buffer stays small because it is broadcast whole every step.
        self.capacity = _HDR + 13 * S + 2 * P + S * MB + 64
        self.buf = torch.zeros(self.capacity, dtype=torch.int32,
                               device=device)
        self._stage = np.zeros(self.capacity, dtype=np.int32)

    # ------------------------------------------------------------------ #
    def _sample_block(self, w, plan) -> None:
        sample = plan.get("sample") or []
        n = len(sample)
        w.put([n])
        if n:
            w.put(_f2i([s[0] for s in sample]))          # temperature
            w.put(np.asarray([s[1] for s in sample], dtype=np.int32))  # top_k
            w.put(_f2i([s[2] for s in sample]))          # top_p
            w.put(np.asarray([s[3] for s in sample], dtype=np.int32))  # seed

    @staticmethod
    def _read_sample(r):
        n = int(r.take(1)[0])
        if not n:
            return []
        temps = _i2f(r.take(n))
        ks = r.take(n)
        ps = _i2f(r.take(n))
        seeds = r.take(n)
        return [(float(temps[i]), int(ks[i]), float(ps[i]), int(seeds[i]))
                for i in range(n)]

    # ------------------------------------------------------------------ #
    def encode(self, plan: Optional[Dict[str, Any]]) -> Optional[torch.Tensor]:
        """Pack a plan dict into the staging buffer; returns the device
        tensor to broadcast, or None when the plan needs the object
        fallback."""
        w = _Writer(self._stage)
        mode = None if plan is None else plan.get("mode")
        try:
            if plan is None or mode == "stop":
                w.put([MODE_STOP, 0])
            elif mode == "decode":
                b = len(plan["tokens"])
                blocks = plan["blocks"]
                w.put([MODE_DECODE, b])
                w.put(np.asarray(plan["tokens"], dtype=np.int32))
                w.put(np.asarray(plan["positions"], dtype=np.int32))
                w.put(np.asarray(plan["slots"], dtype=np.int32))
                w.put(np.asarray(plan["seq_lens"], dtype=np.int32))
                w.put(np.asarray([len(bl) for bl in blocks], dtype=np.int32))
                w.put(np.fromiter((x for bl in blocks for x in bl),
                                  dtype=np.int32))
                self._sample_block(w, plan)
            elif mode == "prefill":
                prompts = plan["prompts"]
                b = len(prompts)
                w.put([MODE_PREFILL, b])
                w.put(np.asarray([len(p) for p in prompts], dtype=np.int32))
                w.put(np.fromiter((x for p in prompts for x in p),
                                  dtype=np.int32))
                w.put(np.fromiter((x for sl in plan["slots"] for x in sl),
                                  dtype=np.int32))
                self._sample_block(w, plan)
            elif mode == "chunk":
                toks = plan["tokens"]
                b = len(toks)
                w.put([MODE_CHUNK, b])
                w.put(np.asarray([len(t) for t in toks], dtype=np.int32))
                w.put(np.fromiter((x for t in toks for x in t),
                                  dtype=np.int32))
                w.put(np.fromiter((x for sl in plan["slots"] for x in sl),
                                  dtype=np.int32))
                w.put(np.asarray(plan["starts"], dtype=np.int32))
                w.put(np.asarray(plan["kv_lens"], dtype=np.int32))
                w.put(np.asarray([1 if c else 0 for c in plan["complete"]],
                                 dtype=np.int32))
                blocks = plan["blocks"]
                w.put(np.asarray([len(bl) for bl in blocks], dtype=np.int32))
                w.put(np.fromiter((x for bl in blocks for x in bl),
                                  dtype=np.int32))
                self._sample_block(w, plan)
            elif mode == "embed":
                prompts = plan["prompts"]
                b = len(prompts)
                w.put([MODE_EMBED, b,
                       1 if plan.get("normalize", True) else 0])
                w.put(np.asarray([len(p) for p in prompts], dtype=np.int32))
                w.put(np.fromiter((x for p in prompts for x in p),
                                  dtype=np.int32))
            else:
                raise ValueError("unknown plan mode {}".format(mode))
        except _Overflow:
            return None
        # no zero-fill of the tail: decode() reads only the declared counts,
        # so stale words beyond w.pos are never interpreted
        src = torch.from_numpy(self._stage)
        if self.device.type == "cuda":
            self.buf.copy_(src, non_blocking=True)
        else:
            self.buf.copy_(src)
        return self.buf

    def decode(self, buf: torch.Tensor) -> Optional[Dict[str, Any]]:
        arr = buf.cpu().numpy() if buf.is_cuda else buf.numpy()
        r = _Reader(arr)
        mode, b = int(arr[0]), int(arr[1])
        r.pos = 2
        if mode == MODE_STOP:
            return {"mode": "stop"}
        if mode == MODE_DECODE:
            tokens = r.take(b).tolist()
            positions = r.take(b).tolist()
            slots = r.take(b).tolist()
            seq_lens = r.take(b).tolist()
            lens = r.take(b)
            blocks = _split(r.take(int(lens.sum())), lens)
            sample = self._read_sample(r)
            return {"mode": "decode", "tokens": tokens, "positions": positions,
                    "slots": slots, "seq_lens": seq_lens, "blocks": blocks,
                    "sample": sample}
        if mode == MODE_PREFILL:
            lens = r.take(b)
            total = int(lens.sum())
            prompts = _split(r.take(total), lens)
            slots = _split(r.take(total), lens)
            sample = self._read_sample(r)
            return {"mode": "prefill", "prompts": prompts, "slots": slots,
                    "sample": sample}
        if mode == MODE_CHUNK:
            lens = r.take(b)
            total = int(lens.sum())
            tokens = _split(r.take(total), lens)
            slots = _split(r.take(total), lens)
            starts = r.take(b).tolist()
            kv_lens = r.take(b).tolist()
            complete = [bool(c) for c in r.take(b)]
            blens = r.take(b)
            blocks = _split(r.take(int(blens.sum())), blens)
            sample = self._read_sample(r)
            return {"mode": "chunk", "tokens": tokens, "slots": slots,
                    "starts": starts, "kv_lens": kv_lens,
                    "complete": complete, "blocks": blocks, "sample": sample}
        if mode == MODE_EMBED:
            normalize = bool(arr[2])
            r.pos = 3
            lens = r.take(b)
            prompts = _split(r.take(int(lens.sum())), lens)
            return {"mode": "embed", "prompts": prompts,
                    "normalize": normalize}
        if mode == MODE_OBJECT:
            return None  # caller falls back to broadcast_object_list
        raise ValueError("bad plan buffer (mode={})".format(mode))

    def mark_object(self) -> torch.Tensor:
        self._stage[:_HDR] = 0
        self._stage[0] = MODE_OBJECT
        self.buf.copy_(torch.from_numpy(self._stage))
        return self.buf


def _split(flat: np.ndarray, lens: np.ndarray):
    """Flattened payload -> per-row python lists (cheap view slicing)."""
    out = []
    pos = 0
    fl = flat.tolist()
    for n in lens.tolist():
        out.append(fl[pos:pos + n])
        pos += n
    return out


class _Overflow(Exception):
    pass


class _Writer:
    def __init__(self, stage: np.ndarray):
        self.stage = stage
        self.pos = 0

    def put(self, vals) -> None:
        arr = np.asarray(vals, dtype=np.int32)
        end = self.pos + arr.size
        if end > self.stage.size:
            raise _Overflow()
        self.stage[self.pos:end] = arr
        self.pos = end


class _Reader:
    def __init__(self, arr: np.ndarray):
        self.arr = arr
        self.pos = 0

    def take(self, n: int) -> np.ndarray:
        out = self.arr[self.pos:self.pos + n]
        self.pos += n
        return out
