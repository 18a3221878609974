"""Llama-3-family dense transformer, built on hypha_amd.ops.

From-scratch modules (no HF): RMSNorm + RoPE + GQA causal attention +
SwiGLU MLP + tied/untied LM head + fused CE loss. Mirrors the model
capability the reference reaches through HF Auto classes
(/root/reference/executors/accelerate/src/hypha/accelerate_executor/model.py),
but implemented directly so the hot ops are our CDNA4 HIP kernels.
"""

from __future__ import annotations

import math
from dataclasses import dataclass, field

import torch
import torch.nn as nn

from hypha_amd import ops


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    hidden_size: int = 4096
    n_layers: int = 32
    n_heads: int = 32
    n_kv_heads: int = 8
    ffn_hidden: int = 14336
    max_seq_len: int = 8192
    rope_base: float = 500000.0
    norm_eps: float = 1e-5
    tie_embeddings: bool = False
    gradient_checkpointing: bool = False
    init_std: float = 0.02
    extra: dict = field(default_factory=dict)

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.n_heads

    def num_params(self) -> int:
        h, v, f = self.hidden_size, self.vocab_size, self.ffn_hidden
        hd = self.head_dim
        attn = h * (self.n_heads * hd) + 2 * h * (self.n_kv_heads * hd) + (self.n_heads * hd) * h
        mlp = 3 * h * f
        per_layer = attn + mlp + 2 * h
        emb = v * h * (1 if self.tie_embeddings else 2)
        return per_layer * self.n_layers + emb + h


# Named presets (sizes per the public Llama-3 architecture; BASELINE.json configs)
PRESETS: dict[str, LlamaConfig] = {
    "llama3-8b": LlamaConfig(),
    "llama3-70b": LlamaConfig(
        hidden_size=8192, n_layers=80, n_heads=64, n_kv_heads=8, ffn_hidden=28672
    ),
    # small debug model for CPU tests
    "llama-tiny": LlamaConfig(
        vocab_size=512, hidden_size=128, n_layers=2, n_heads=2, n_kv_heads=2,
        ffn_hidden=256, max_seq_len=256, rope_base=10000.0,
    ),
    # ~124M GPT-2-small-scale llama-style model for the CPU plumbing config
    "gpt2-small": LlamaConfig(
        vocab_size=50304, hidden_size=768, n_layers=12, n_heads=12, n_kv_heads=12,
        ffn_hidden=2048, max_seq_len=1024, rope_base=10000.0, tie_embeddings=True,
    ),
}


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, x):
        return ops.rmsnorm(x, self.weight, self.eps)


class Attention(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        hd = cfg.head_dim
        self.wq = nn.Linear(cfg.hidden_size, cfg.n_heads * hd, bias=False)
        self.wk = nn.Linear(cfg.hidden_size, cfg.n_kv_heads * hd, bias=False)
        self.wv = nn.Linear(cfg.hidden_size, cfg.n_kv_heads * hd, bias=False)
        self.wo = nn.Linear(cfg.n_heads * hd, cfg.hidden_size, bias=False)

    def forward(self, x, cos, sin, cache=None, pos: int = 0):
        b, s, _ = x.shape
        cfg = self.cfg
        hd = cfg.head_dim
        # BSHD layout throughout: the projections' natural layout, consumed
        # directly by the stride-aware RoPE/attention kernels (no transposes)
        q = self.wq(x).view(b, s, cfg.n_heads, hd)
        k = self.wk(x).view(b, s, cfg.n_kv_heads, hd)
        v = self.wv(x).view(b, s, cfg.n_kv_heads, hd)
        if cache is None:
            q, k = ops.apply_rope_qk(q, k, cos, sin, layout="bshd")
            o = ops.flash_attention(q, k, v, causal=True, layout="bshd")
            return self.wo(o.reshape(b, s, -1))
        # inference with KV cache: rotate at absolute positions, append into
        # the pre-allocated bshd cache, then attend over the valid prefix.
        # Decode (s==1) runs the native flash-decoding kernel; prefill runs
        # the MFMA flash kernel; chunked continuation falls back to fp32.
        q, k = ops.apply_rope_qk(q, k, cos[pos:], sin[pos:], layout="bshd")
        t = cache.append(k, v)
        if s == 1:
            o = ops.attn_decode(q[:, 0].contiguous(), cache.k, cache.v, t,
                                cache.k_scale, cache.v_scale)
            return self.wo(o.reshape(b, 1, -1))
        if pos == 0 and (not x.is_cuda or s % 128 == 0):
            o = ops.flash_attention(q, k, v, causal=True, layout="bshd")
            return self.wo(o.reshape(b, s, -1))
        # chunked prefill with a position offset (rare path; fp32 reference)
        kc, vc = cache.dequant(t)
        rep = cfg.n_heads // cfg.n_kv_heads
        qh = q.transpose(1, 2)  # [b, hq, s, d]
        kh = kc.transpose(1, 2).repeat_interleave(rep, dim=1)
        vh = vc.transpose(1, 2).repeat_interleave(rep, dim=1)
        scores = (qh.float() @ kh.float().transpose(-1, -2)) / math.sqrt(hd)
        mask = torch.arange(t, device=x.device)[None, :] > (
            pos + torch.arange(s, device=x.device)[:, None]
        )
        scores = scores.masked_fill(mask, float("-inf"))
        p = torch.softmax(scores, dim=-1)
        o = (p @ vh.float()).to(x.dtype).transpose(1, 2)
        return self.wo(o.reshape(b, s, -1))


class MLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.w_gate = nn.Linear(cfg.hidden_size, cfg.ffn_hidden, bias=False)
        self.w_up = nn.Linear(cfg.hidden_size, cfg.ffn_hidden, bias=False)
        self.w_down = nn.Linear(cfg.ffn_hidden, cfg.hidden_size, bias=False)

    def forward(self, x):
        return self.w_down(ops.swiglu(self.w_gate(x), self.w_up(x)))


class Block(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.attn_norm = RMSNorm(cfg.hidden_size, cfg.norm_eps)
        self.attn = Attention(cfg)
        self.mlp_norm = RMSNorm(cfg.hidden_size, cfg.norm_eps)
        self.mlp = MLP(cfg)

    def forward(self, x, cos, sin, cache=None, pos: int = 0):
        attn_out = self.attn(self.attn_norm(x), cos, sin, cache=cache, pos=pos)
        # fused residual-add + norm: one kernel writes the residual stream
        # and the normalized MLP input (saves an elementwise pass per block)
        x, n2 = ops.add_rmsnorm(x, attn_out, self.mlp_norm.weight,
                                self.mlp_norm.eps)
        x = x + self.mlp(n2)
        return x

    def forward_pair(self, res, delta, cos, sin):
        """Training fast path with the residual add DEFERRED: receives the
        previous block's (residual, sub-block output) pair so EVERY residual
        add fuses into the next norm kernel (including across blocks)."""
        res, n1 = ops.add_rmsnorm(res, delta, self.attn_norm.weight,
                                  self.attn_norm.eps)
        attn_out = self.attn(n1, cos, sin)
        res, n2 = ops.add_rmsnorm(res, attn_out, self.mlp_norm.weight,
                                  self.mlp_norm.eps)
        return res, self.mlp(n2)


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.blocks = nn.ModuleList(Block(cfg) for _ in range(cfg.n_layers))
        self.norm = RMSNorm(cfg.hidden_size, cfg.norm_eps)
        if cfg.tie_embeddings:
            self.lm_head = None
        else:
            self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        cos, sin = ops.reference.rope_cos_sin(cfg.max_seq_len, cfg.head_dim, cfg.rope_base)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        self.apply(self._init)
        # scaled init on residual-out projections (GPT-2/Llama practice)
        for blk in self.blocks:
            for lin in (blk.attn.wo, blk.mlp.w_down):
                nn.init.normal_(lin.weight, std=cfg.init_std / math.sqrt(2 * cfg.n_layers))

    def _init(self, m):
        if isinstance(m, nn.Linear):
            nn.init.normal_(m.weight, std=self.cfg.init_std)
        elif isinstance(m, nn.Embedding):
            nn.init.normal_(m.weight, std=self.cfg.init_std)

    def forward(self, input_ids: torch.Tensor, labels: torch.Tensor | None = None):
        x = self.embed(input_ids)
        cos, sin = self.rope_cos, self.rope_sin
        # (residual, delta) carry: every residual add — including the
        # cross-block one — fuses into the next RMSNorm kernel; the final
        # add fuses into the output norm
        delta = torch.zeros_like(x)
        for blk in self.blocks:
            if self.cfg.gradient_checkpointing and self.training:
                x, delta = torch.utils.checkpoint.checkpoint(
                    blk.forward_pair, x, delta, cos, sin, use_reentrant=False)
            else:
                x, delta = blk.forward_pair(x, delta, cos, sin)
        _, x = ops.add_rmsnorm(x, delta, self.norm.weight, self.norm.eps)
        if self.lm_head is not None:
            logits = self.lm_head(x)
        else:
            logits = torch.nn.functional.linear(x, self.embed.weight)
        if labels is None:
            return logits
        # next-token prediction: shift
        loss = ops.cross_entropy_loss(logits[:, :-1], labels[:, 1:])
        return loss


@torch.no_grad()
def _generate(self, input_ids: torch.Tensor, max_new_tokens: int = 32,
              temperature: float = 0.0, top_k: int = 0,
              seed: int | None = None, kv_quant: str | None = None) -> torch.Tensor:
    """Autoregressive generation with a per-layer KV cache (inference parity:
    the reference serves inference through the same executor surface)."""
    from .kv_cache import KVCache

    self.eval()
    b, s = input_ids.shape
    max_len = s + max_new_tokens
    dtype = self.embed.weight.dtype
    caches = [KVCache(b, max_len, self.cfg.n_kv_heads, self.cfg.head_dim,
                      input_ids.device, dtype=dtype, quant=kv_quant)
              for _ in self.blocks]
    gen = None
    if seed is not None:
        gen = torch.Generator(device=input_ids.device).manual_seed(seed)
    tokens = input_ids
    x_in = input_ids
    pos = 0
    for _ in range(max_new_tokens):
        x = self.embed(x_in)
        for blk, cache in zip(self.blocks, caches):
            x = blk(x, self.rope_cos, self.rope_sin, cache=cache, pos=pos)
        x = self.norm(x[:, -1:])
        if self.lm_head is not None:
            logits = self.lm_head(x)[:, 0]
        else:
            logits = torch.nn.functional.linear(x, self.embed.weight)[:, 0]
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


LlamaForCausalLM.generate = _generate


def build_model(name: str, **overrides) -> LlamaForCausalLM:
    cfg = PRESETS[name]
    if overrides:
        from dataclasses import replace

        cfg = replace(cfg, **overrides)
    return LlamaForCausalLM(cfg)
