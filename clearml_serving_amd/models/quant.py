"""fp8 (OCP e4m3) weight quantization for LLM serving.

Decode is weight-streaming bound: fp8 weights halve the HBM bytes per step.
Round 1 ran fp8 through hipBLASLt with FOUR un-fused elementwise passes of
dynamic activation quantization per projection and measured SLOWER than
bf16 end to end (3.7k vs 6.0k out-tok/s). Round 2 fuses the quantization
into the producing kernels -- ops.rmsnorm_fp8 / ops.silu_mul_fp8 emit fp8
bytes + per-row scales directly (one HBM pass), and decode-shaped GEMMs run
on the in-tree fp8 MFMA skinny kernel (ops.skinny_gemm_fp8) that fills the
256-CU chip where hipBLASLt's tiles cannot. Prefill-shaped GEMMs stay on
hipBLASLt scaled_mm (measured 1.69x vs bf16 at M=2048).

Scheme: per-output-row weight scales (absmax/448), dynamic per-row (token)
activation scales. Enable per endpoint via model card / aux
``{"quantization": "fp8"}``. The reference delegates all quantization to
vLLM (preprocess_service.py:619-1095).
"""

import torch
import torch.nn as nn

from .. import ops

F8 = torch.float8_e4m3fn
F8_MAX = 448.0


class Fp8Linear(nn.Module):
    """nn.Linear replacement: fp8 weights with per-row scales (an optional
    bias -- Qwen2 QKV -- stays in the compute dtype, added after the GEMM).

    Two entry points:
      forward(x)           -- bf16 in: quantize (one fused kernel) then GEMM
      forward_q(x8, xs)    -- pre-quantized activations from a fused
                              producer (rmsnorm_fp8 / silu_mul_fp8)
    """

    # decode-shaped rows route to the in-tree fp8 MFMA kernel; above this
    # hipBLASLt scaled_mm wins (measured crossover, profiles/fp8_kernels)
    SKINNY_MAX_M = int(__import__("os").environ.get("CMLS_FP8_SKINNY_MAX", 16))

    def __init__(self, linear: nn.Linear):
        super().__init__()
        w = linear.weight.data.float()
        row_max = w.abs().amax(dim=1, keepdim=True).clamp(min=1e-8)
        scale = row_max / F8_MAX
        w8 = (w / scale).to(F8).contiguous()
        self.register_buffer("weight_fp8", w8)
        # decode kernel reads a byte-swizzled copy (one 16-B lane load =
        # both MFMA operands of a 64-wide k pair); prefill scaled_mm needs
        # the plain layout, so both live in HBM (fp8: still half of bf16)
        self.register_buffer("weight_fp8_sw", ops.swizzle_fp8_weight(w8))
        self.register_buffer("weight_scale",
                             scale.to(torch.float32).reshape(-1)
                             .contiguous())  # [N]
        self.out_features = linear.out_features
        self.in_features = linear.in_features
        self.out_dtype = linear.weight.dtype  # model compute dtype
        # Qwen2 QKV bias: added in the compute dtype after the fp8 GEMM
        if linear.bias is not None:
            self.register_buffer("bias", linear.bias.data.clone())
        else:
            self.bias = None

    # LDS-staged v2 (skinny_gemm_fp8_v2) beats STANDALONE hipBLASLt
    # scaled_mm at M 17..64 on qkv/o_proj shapes, but INSIDE the decode
    # hipGraphs scaled_mm has no launch floor and wins end to end
    # (12,170 vs 12,076 out-tok/s at 64 seqs, 6,928 vs 6,230 at 32) --
    # so v2 is OFF by default and kept as the measured exhibit
    # (CMLS_FP8_V2_MAX raises the cap for standalone/no-graph serving).
    V2_MAX_M = int(__import__("os").environ.get("CMLS_FP8_V2_MAX", 0))

    def _route(self, m: int) -> str:
        """Measured per-shape routing (profiles/fp8_kernels.txt):
        - v1 (8-wave k-split, swizzled weights) owns M<=16 -- except the
          huge-N gate_up shape above M=1, where hipBLASLt's ~22 us floor
          wins
        - v2 (LDS-staged, 4 waves) owns M 17..64 on the qkv/o_proj-class
          shapes (N < 16k, K < 8k)
        - hipBLASLt scaled_mm keeps huge-N / huge-K shapes at M >= 17 and
          everything above 64."""
        n, k = self.out_features, self.in_features
        if not (k % 64 == 0 and n % 16 == 0):
            return "lt"
        if m <= self.SKINNY_MAX_M:
            if n >= 16384 and m > 1:
                return "lt"
            return "v1"
        if m <= self.V2_MAX_M and k % 128 == 0 and n < 16384 and k < 8192:
            return "v2"
        return "lt"

    def forward_q(self, x8: torch.Tensor, x_scale: torch.Tensor,
                  out_dtype=None) -> torch.Tensor:
        out_dtype = out_dtype or self.out_dtype
        out = self._gemm_q(x8, x_scale, out_dtype)
        if self.bias is not None:
            out = out + self.bias.to(out_dtype)
        return out

    def _gemm_q(self, x8: torch.Tensor, x_scale: torch.Tensor,
                out_dtype) -> torch.Tensor:
        m = x8.shape[0]
        route = self._route(m) if x8.is_cuda else "lt"
        if route == "v1":
            out = ops.skinny_gemm_fp8(x8, x_scale, self.weight_fp8_sw,
                                      self.weight_scale, swizzled=True)
            return out if out.dtype == out_dtype else out.to(out_dtype)
        if route == "v2":
            out = ops.skinny_gemm_fp8_v2(x8, x_scale, self.weight_fp8,
                                         self.weight_scale)
            return out if out.dtype == out_dtype else out.to(out_dtype)
        if not x8.is_cuda:
            return ops.skinny_gemm_fp8(x8, x_scale, self.weight_fp8,
                                       self.weight_scale).to(out_dtype)
        # prefill-shaped: hipBLASLt scaled_mm (rowwise scales); M % 16 pad
        pad = (-m) % 16
        xq = x8.view(F8)
        xs = x_scale.reshape(-1, 1)
        if pad:
            xq = torch.nn.functional.pad(xq.view(torch.uint8),
                                         (0, 0, 0, pad)).view(F8)
            xs = torch.nn.functional.pad(xs, (0, 0, 0, pad), value=1.0)
        out = torch._scaled_mm(
            xq, self.weight_fp8.t(), scale_a=xs.contiguous(),
            scale_b=self.weight_scale.reshape(1, -1), out_dtype=out_dtype)
        return out[:m] if pad else out

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        orig_shape = x.shape
        x2 = x.reshape(-1, orig_shape[-1])
        x8, xs = ops.quant_fp8(x2)
        out = self.forward_q(x8, xs, out_dtype=x.dtype)
        return out.reshape(*orig_shape[:-1], self.out_features)


def quantize_llama_fp8(model: nn.Module) -> int:
    """Swap the projection Linears of a LlamaForCausalLM to Fp8Linear
    (lm_head stays bf16 for logit fidelity). Returns layers converted."""
    n = 0
    for layer in getattr(model, "layers", []):
        for name in ("qkv", "o_proj", "gate_up", "down"):
            lin = getattr(layer, name, None)
            if isinstance(lin, nn.Linear):
                setattr(layer, name, Fp8Linear(lin))
                n += 1
    return n
