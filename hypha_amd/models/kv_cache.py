"""Pre-allocated per-layer KV cache for autoregressive serving.

bshd layout [B, T_max, Hkv, D] — the projections' natural output layout, so
appends are plain slice copies (no per-step torch.cat reallocation, which is
O(T^2) bytes over a decode) and the flash-decoding kernel
(ops/hip/decode.hip) reads it in place.
"""

from __future__ import annotations

import torch


class KVCache:
    __slots__ = ("k", "v", "t")

    def __init__(self, batch: int, max_len: int, n_kv_heads: int, head_dim: int,
                 device, dtype=torch.bfloat16):
        self.k = torch.zeros(batch, max_len, n_kv_heads, head_dim,
                             device=device, dtype=dtype)
        self.v = torch.zeros_like(self.k)
        self.t = 0

    def append(self, k: torch.Tensor, v: torch.Tensor) -> int:
        """k/v [B, s, Hkv, D]; returns the new valid length."""
        s = k.shape[1]
        if self.t + s > self.k.shape[1]:
            raise ValueError(
                f"KV cache overflow: {self.t}+{s} > {self.k.shape[1]}")
        self.k[:, self.t:self.t + s] = k
        self.v[:, self.t:self.t + s] = v
        self.t += s
        return self.t
