"""GPT-2 architecture (LayerNorm + learned positions + GELU MLP), from scratch.

This is the BASELINE.json config-1 plumbing model ("GPT-2-small DiLoCo H=10,
2 CPU worker processes on gloo"): faithful GPT-2 wiring so the framework's
model registry covers the reference's GPT2 ModelType
(/root/reference/crates/messages/src/lib.rs:419-488 ModelType list) without
pulling HF weights. Runs on the reference op path on CPU and on library GEMMs
+ our attention kernel on GPU.
"""

from __future__ import annotations

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from hypha_amd import ops


@dataclass
class GPT2Config:
    vocab_size: int = 50304  # 50257 rounded up for GEMM-friendly vocab
    n_positions: int = 1024
    hidden_size: int = 768
    n_layers: int = 12
    n_heads: int = 12
    norm_eps: float = 1e-5
    init_std: float = 0.02


PRESETS = {
    "gpt2-small": GPT2Config(),
    "gpt2-tiny": GPT2Config(vocab_size=512, n_positions=128, hidden_size=64, n_layers=2, n_heads=4),
}


class LayerNorm(nn.LayerNorm):
    """nn.LayerNorm parameters, dispatched to the native HIP kernel on GPU
    (ops/hip/layernorm.hip) and torch on CPU."""

    def forward(self, x):
        return ops.layernorm(x, self.weight, self.bias, self.eps)


class GPT2Block(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        h = cfg.hidden_size
        self.ln1 = LayerNorm(h, eps=cfg.norm_eps)
        self.attn_qkv = nn.Linear(h, 3 * h)
        self.attn_out = nn.Linear(h, h)
        self.ln2 = LayerNorm(h, eps=cfg.norm_eps)
        self.mlp_fc = nn.Linear(h, 4 * h)
        self.mlp_proj = nn.Linear(4 * h, h)
        self.n_heads = cfg.n_heads

    def forward(self, x, cache=None, pos: int = 0):
        b, s, h = x.shape
        hd = h // self.n_heads
        q, k, v = self.attn_qkv(self.ln1(x)).split(h, dim=-1)
        if cache is None:
            q = q.view(b, s, self.n_heads, hd).transpose(1, 2)
            k = k.view(b, s, self.n_heads, hd).transpose(1, 2)
            v = v.view(b, s, self.n_heads, hd).transpose(1, 2)
            o = ops.flash_attention(q, k, v, causal=True)
            x = x + self.attn_out(o.transpose(1, 2).reshape(b, s, h))
        else:
            # serving: append into the pre-allocated bshd cache; decode via
            # the flash-decoding kernel, prefill via the MFMA flash kernel
            q = q.view(b, s, self.n_heads, hd)
            k = k.view(b, s, self.n_heads, hd)
            v = v.view(b, s, self.n_heads, hd)
            t = cache.append(k, v)
            if s == 1:
                o = ops.attn_decode(q[:, 0].contiguous(), cache.k, cache.v, t,
                                    cache.k_scale, cache.v_scale)
                o = o.reshape(b, 1, h)
            elif pos == 0 and (not x.is_cuda or s % 128 == 0):
                o = ops.flash_attention(q, k, v, causal=True, layout="bshd")
                o = o.reshape(b, s, h)
            else:  # ragged prefill / chunked continuation (fp32 reference)
                kd, vd = cache.dequant(t)
                kh = kd.transpose(1, 2)
                vh = vd.transpose(1, 2)
                qh = q.transpose(1, 2)
                scores = (qh.float() @ kh.float().transpose(-1, -2)) / math.sqrt(hd)
                mask = torch.arange(t, device=x.device)[None, :] > (
                    pos + torch.arange(s, device=x.device)[:, None])
                p = torch.softmax(scores.masked_fill(mask, float("-inf")), dim=-1)
                o = (p @ vh.float()).to(x.dtype).transpose(1, 2).reshape(b, s, h)
            x = x + self.attn_out(o)
        x = x + self.mlp_proj(ops.gelu(self.mlp_fc(self.ln2(x))))
        return x


class GPT2ForCausalLM(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.cfg = cfg
        self.wte = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.wpe = nn.Embedding(cfg.n_positions, cfg.hidden_size)
        self.blocks = nn.ModuleList(GPT2Block(cfg) for _ in range(cfg.n_layers))
        self.ln_f = LayerNorm(cfg.hidden_size, eps=cfg.norm_eps)
        self.apply(self._init)
        for blk in self.blocks:
            for lin in (blk.attn_out, blk.mlp_proj):
                nn.init.normal_(lin.weight, std=cfg.init_std / math.sqrt(2 * cfg.n_layers))

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=self.cfg.init_std)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward(self, input_ids: torch.Tensor, labels: torch.Tensor | None = None):
        b, s = input_ids.shape
        pos = torch.arange(s, device=input_ids.device)
        x = self.wte(input_ids) + self.wpe(pos)
        for blk in self.blocks:
            x = blk(x)
        x = self.ln_f(x)
        logits = F.linear(x, self.wte.weight)  # tied head
        if labels is None:
            return logits
        return ops.cross_entropy_loss(logits[:, :-1], labels[:, 1:])

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int = 32,
                 temperature: float = 0.0, top_k: int = 0,
                 seed: int | None = None) -> torch.Tensor:
        """KV-cache decoding (prefill once, then one token per step via the
        flash-decoding kernel); same sampling surface as the Llama family so
        inference jobs can run any registry model. Sequences are capped at
        n_positions (absolute position embeddings)."""
        from .kv_cache import KVCache

        self.eval()
        b, s = input_ids.shape
        max_len = min(s + max_new_tokens, self.cfg.n_positions)
        hd = self.cfg.hidden_size // self.cfg.n_heads
        dtype = self.wte.weight.dtype
        caches = [KVCache(b, max_len, self.cfg.n_heads, hd, input_ids.device,
                          dtype=dtype) for _ in self.blocks]
        gen = None
        if seed is not None:
            gen = torch.Generator(device=input_ids.device).manual_seed(seed)
        tokens = input_ids
        x_in = input_ids
        pos = 0
        for _ in range(max_new_tokens):
            if pos + x_in.shape[1] > self.cfg.n_positions:
                break  # absolute-position ceiling reached
            idx = torch.arange(pos, pos + x_in.shape[1], device=x_in.device)
            x = self.wte(x_in) + self.wpe(idx)
            for blk, cache in zip(self.blocks, caches):
                x = blk(x, cache=cache, pos=pos)
            x = self.ln_f(x[:, -1:])
            logits = F.linear(x, self.wte.weight)[:, 0]
            if temperature <= 0:
                nxt = logits.argmax(-1, keepdim=True)
            else:
                logits = logits / temperature
                if top_k > 0:
                    kth = logits.topk(top_k, dim=-1).values[..., -1, None]
                    logits = logits.masked_fill(logits < kth, float("-inf"))
                probs = torch.softmax(logits.float(), dim=-1)
                nxt = torch.multinomial(probs, 1, generator=gen)
            pos += x_in.shape[1]
            tokens = torch.cat([tokens, nxt], dim=1)
            x_in = nxt
        return tokens


def build_model(name: str) -> GPT2ForCausalLM:
    return GPT2ForCausalLM(PRESETS[name])
