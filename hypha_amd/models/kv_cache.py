"""Pre-allocated per-layer KV cache for autoregressive serving.

bshd layout [B, T_max, Hkv, D] — the projections' natural output layout, so
appends are plain slice copies (no per-step torch.cat reallocation, which is
O(T^2) bytes over a decode) and the flash-decoding kernel
(ops/hip/decode.hip) reads it in place.

Optional fp8 mode (quant="fp8"): rows are stored as OCP e4m3 with one fp32
scale per (batch, position, kv-head) row — appends quantize with their own
row amax so history never needs requantization, cache memory and decode
cache reads halve, and the decode kernel dequantizes during LDS staging.
"""

from __future__ import annotations

import torch

E4M3_MAX = 448.0


class KVCache:
    __slots__ = ("k", "v", "t", "quant", "k_scale", "v_scale")

    def __init__(self, batch: int, max_len: int, n_kv_heads: int, head_dim: int,
                 device, dtype=torch.bfloat16, quant: str | None = None):
        self.quant = quant
        if quant == "fp8":
            self.k = torch.zeros(batch, max_len, n_kv_heads, head_dim,
                                 device=device, dtype=torch.float8_e4m3fn)
            self.v = torch.zeros_like(self.k)
            self.k_scale = torch.ones(batch, max_len, n_kv_heads,
                                      device=device, dtype=torch.float32)
            self.v_scale = torch.ones_like(self.k_scale)
        elif quant is None:
            self.k = torch.zeros(batch, max_len, n_kv_heads, head_dim,
                                 device=device, dtype=dtype)
            self.v = torch.zeros_like(self.k)
            self.k_scale = self.v_scale = None
        else:
            raise ValueError(f"unknown KV quant mode {quant!r}")
        self.t = 0

    def _quantize(self, x: torch.Tensor, scale_out: torch.Tensor) -> torch.Tensor:
        amax = x.float().abs().amax(-1).clamp(min=1e-8)  # [B, s, Hkv]
        scale = amax / E4M3_MAX
        scale_out.copy_(scale)
        return (x.float() / scale.unsqueeze(-1)).clamp(-E4M3_MAX, E4M3_MAX).to(
            torch.float8_e4m3fn)

    def append(self, k: torch.Tensor, v: torch.Tensor) -> int:
        """k/v [B, s, Hkv, D] bf16; returns the new valid length."""
        s = k.shape[1]
        if self.t + s > self.k.shape[1]:
            raise ValueError(
                f"KV cache overflow: {self.t}+{s} > {self.k.shape[1]}")
        if self.quant == "fp8":
            self.k[:, self.t:self.t + s] = self._quantize(
                k, self.k_scale[:, self.t:self.t + s])
            self.v[:, self.t:self.t + s] = self._quantize(
                v, self.v_scale[:, self.t:self.t + s])
        else:
            self.k[:, self.t:self.t + s] = k
            self.v[:, self.t:self.t + s] = v
        self.t += s
        return self.t

    def append_at(self, k: torch.Tensor, v: torch.Tensor, pos: torch.Tensor) -> None:
        """Graph-capturable single-row append at a device position index
        (s == 1); does NOT advance the python-side counter."""
        if self.quant == "fp8":
            if k.is_cuda and k.shape[-1] in (64, 128):
                from hypha_amd import _C

                _C.kv_append_fp8_(k.contiguous(), v.contiguous(), self.k,
                                  self.v, self.k_scale, self.v_scale, pos)
                return
            kq = self._quantize_rows(k)
            vq = self._quantize_rows(v)
            # index_copy_ has no fp8 kernel: operate on the byte view
            self.k.view(torch.uint8).index_copy_(1, pos, kq[0].view(torch.uint8))
            self.v.view(torch.uint8).index_copy_(1, pos, vq[0].view(torch.uint8))
            self.k_scale.index_copy_(1, pos, kq[1])
            self.v_scale.index_copy_(1, pos, vq[1])
        else:
            self.k.index_copy_(1, pos, k)
            self.v.index_copy_(1, pos, v)

    def _quantize_rows(self, x: torch.Tensor):
        amax = x.float().abs().amax(-1).clamp(min=1e-8)
        scale = amax / E4M3_MAX
        q = (x.float() / scale.unsqueeze(-1)).clamp(-E4M3_MAX, E4M3_MAX).to(
            torch.float8_e4m3fn)
        return q, scale

    def dequant(self, t: int | None = None):
        """bf16 views/copies of the valid prefix (fallback paths)."""
        t = self.t if t is None else t
        if self.quant is None:
            return self.k[:, :t], self.v[:, :t]
        kd = self.k[:, :t].float() * self.k_scale[:, :t].unsqueeze(-1)
        vd = self.v[:, :t].float() * self.v_scale[:, :t].unsqueeze(-1)
        return kd.bfloat16(), vd.bfloat16()
