"""FP8 (OCP e4m3) linear layers via hipBLASLt scaled GEMMs — BASELINE
config 5's fp8 compute path. Measured on MI355X: 2566 TF/s vs 1281 TF/s bf16
at M8192 K4096 N4096 (2.0x; CDNA4 fp8 MFMA dense peak is ~5 PF).

Dynamic per-tensor scaling: x and w are scaled to the e4m3 range (max 448)
per call; forward and both backward GEMMs run in fp8 with bf16 outputs.
Weights stay bf16 masters (the DiLoCo flat-parameter flow is unchanged);
fp8 is a compute/datatype transform, not a storage change — fp8 weight
STORAGE for the 70B config is the round-2 memory plan (see docs/memory.md).
"""

from __future__ import annotations

import torch
import torch.nn as nn

E4M3_MAX = 448.0


def _to_fp8(t: torch.Tensor):
    """Dynamic per-tensor symmetric scaling into e4m3; returns (fp8, scale)."""
    amax = t.abs().amax().float().clamp(min=1e-12)
    scale = amax / E4M3_MAX
    t8 = (t.float() / scale).clamp(-E4M3_MAX, E4M3_MAX).to(torch.float8_e4m3fn)
    return t8, scale


def _scaled_mm(a8, b8_colmajor, sa, sb):
    return torch._scaled_mm(a8, b8_colmajor, scale_a=sa, scale_b=sb,
                            out_dtype=torch.bfloat16)


class _Fp8Linear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x2d, weight):
        x8, sx = _to_fp8(x2d)
        w8, sw = _to_fp8(weight)  # [N, K]
        out = _scaled_mm(x8, w8.t(), sx, sw)  # [M, N]
        ctx.save_for_backward(x2d, weight)
        return out

    @staticmethod
    def backward(ctx, dy):
        x2d, weight = ctx.saved_tensors
        dy = dy.contiguous()
        dy8, sdy = _to_fp8(dy)
        # dx = dy @ W : B must be column-major [N, K]
        wt8, swt = _to_fp8(weight.t().contiguous())  # [K, N]
        dx = _scaled_mm(dy8, wt8.t(), sdy, swt)
        # dw = dy^T @ x : A row-major [N, M]; B column-major [M, K]
        dyt8, sdyt = _to_fp8(dy.t().contiguous())
        xt8, sxt = _to_fp8(x2d.t().contiguous())  # [K, M]
        dw = _scaled_mm(dyt8, xt8.t(), sdyt, sxt)
        return dx, dw


class Fp8Linear(nn.Module):
    """Drop-in nn.Linear (no bias) running its GEMMs in fp8 on GPU."""

    def __init__(self, in_features: int, out_features: int):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        nn.init.normal_(self.weight, std=0.02)

    @classmethod
    def from_linear(cls, lin: nn.Linear) -> "Fp8Linear":
        m = cls.__new__(cls)
        nn.Module.__init__(m)
        m.weight = lin.weight
        return m

    def forward(self, x):
        if not x.is_cuda:  # CPU path: plain matmul (tests/plumbing)
            return torch.nn.functional.linear(x, self.weight)
        shape = x.shape
        out = _Fp8Linear.apply(x.reshape(-1, shape[-1]).contiguous(), self.weight)
        return out.reshape(*shape[:-1], -1)


def convert_linears_to_fp8(model: nn.Module, min_features: int = 1024,
                           max_features: int = 65536) -> int:
    """Swap every large nn.Linear (bias-free) for Fp8Linear. Returns count.
    Small projections (routers, tiny models) stay bf16; so does the vocab
    projection (> max_features): its backward would materialize transposed
    fp8 copies of the [tokens, vocab] gradient (gigabytes)."""
    n = 0
    for parent in model.modules():
        for name, child in list(parent.named_children()):
            if (isinstance(child, nn.Linear) and child.bias is None
                    and min_features <= child.in_features <= max_features
                    and min_features <= child.out_features <= max_features):
                setattr(parent, name, Fp8Linear.from_linear(child))
                n += 1
    return n
