"""Mixtral-style sparse MoE transformer (BASELINE config 4: Mixtral 8x7B
DiLoCo, MoE grouped-GEMM inner step, fits 288 GB HBM).

Same attention stack as the Llama family (our CDNA4 kernels); the MLP is a
top-2 router over N experts with token grouping: tokens are sorted by
expert assignment and each expert's tokens run as one dense GEMM group
(hipBLASLt), with our SwiGLU kernel fused in between. The reference only
reaches MoE through HF Auto classes (model.py's 38-way map); here it is a
first-class family.
"""

from __future__ import annotations

from dataclasses import dataclass, field

import torch
import torch.nn as nn
import torch.nn.functional as F

from hypha_amd import ops
from .llama import Attention, LlamaConfig, RMSNorm


@dataclass
class MoEConfig(LlamaConfig):
    n_experts: int = 8
    top_k: int = 2
    router_aux_loss_coef: float = 0.01

    def num_params(self) -> int:
        h, v, f = self.hidden_size, self.vocab_size, self.ffn_hidden
        hd = self.head_dim
        attn = h * (self.n_heads * hd) + 2 * h * (self.n_kv_heads * hd) + (self.n_heads * hd) * h
        mlp = 3 * h * f * self.n_experts + h * self.n_experts
        per_layer = attn + mlp + 2 * h
        emb = v * h * (1 if self.tie_embeddings else 2)
        return per_layer * self.n_layers + emb + h


PRESETS: dict[str, MoEConfig] = {
    # Mixtral-8x7B architecture (public config)
    "mixtral-8x7b": MoEConfig(
        vocab_size=32000, hidden_size=4096, n_layers=32, n_heads=32, n_kv_heads=8,
        ffn_hidden=14336, max_seq_len=8192, rope_base=1e6, n_experts=8, top_k=2,
    ),
    "moe-tiny": MoEConfig(
        vocab_size=512, hidden_size=128, n_layers=2, n_heads=2, n_kv_heads=2,
        ffn_hidden=256, max_seq_len=256, rope_base=10000.0, n_experts=4, top_k=2,
    ),
}


def _a2a_exchange(out: torch.Tensor, x: torch.Tensor, out_splits, in_splits, group):
    """all_to_all_single with a gloo fallback (gloo has no all-to-all:
    emulated with batched isend/irecv pairs; self-chunk is a local copy)."""
    import torch.distributed as dist

    if dist.get_backend(group) != "gloo":
        dist.all_to_all_single(out, x, out_splits, in_splits, group=group)
        return
    rank = dist.get_rank(group)
    size = dist.get_world_size(group)
    in_off = [0]
    for c in in_splits:
        in_off.append(in_off[-1] + c)
    out_off = [0]
    for c in out_splits:
        out_off.append(out_off[-1] + c)
    reqs = []
    for r in range(size):
        if r == rank:
            out[out_off[r]:out_off[r + 1]] = x[in_off[r]:in_off[r + 1]]
            continue
        if in_splits[r]:
            reqs.append(dist.isend(x[in_off[r]:in_off[r + 1]].contiguous(),
                                   dist.get_global_rank(group, r), group=group))
        if out_splits[r]:
            reqs.append(dist.irecv(out[out_off[r]:out_off[r + 1]],
                                   dist.get_global_rank(group, r), group=group))
    for q in reqs:
        q.wait()


class _AllToAll(torch.autograd.Function):
    """Autograd token exchange: backward is the transposed all-to-all."""

    @staticmethod
    def forward(ctx, x, out_splits, in_splits, group):
        ctx.group = group
        ctx.out_splits = out_splits
        ctx.in_splits = in_splits
        out = x.new_empty((sum(out_splits),) + tuple(x.shape[1:]))
        _a2a_exchange(out, x.contiguous(), out_splits, in_splits, group)
        return out

    @staticmethod
    def backward(ctx, g):
        out = g.new_empty((sum(ctx.in_splits),) + tuple(g.shape[1:]))
        _a2a_exchange(out, g.contiguous(), ctx.in_splits, ctx.out_splits, ctx.group)
        return out, None, None, None


class MoEMLP(nn.Module):
    """Top-k routed expert MLP with token grouping (grouped-GEMM execution).

    Optional EXPERT PARALLELISM (shard_experts_): experts are sharded across
    an RCCL group; tokens travel to their experts by all-to-all over xGMI
    and return after the expert GEMMs. Router and attention stay replicated
    (DiLoCo-synced); expert weights are singletons (each update sees every
    rank's tokens), so they are excluded from the outer all-reduce."""

    def __init__(self, cfg: MoEConfig):
        super().__init__()
        self.cfg = cfg
        h, f, e = cfg.hidden_size, cfg.ffn_hidden, cfg.n_experts
        self.router = nn.Linear(h, e, bias=False)
        self.w_gate = nn.Parameter(torch.empty(e, f, h))
        self.w_up = nn.Parameter(torch.empty(e, f, h))
        self.w_down = nn.Parameter(torch.empty(e, h, f))
        for w in (self.w_gate, self.w_up, self.w_down):
            nn.init.normal_(w, std=cfg.init_std)
        self.ep = None  # (rank, size, group) after shard_experts_

    def forward(self, x):
        cfg = self.cfg
        b, s, h = x.shape
        xf = x.reshape(-1, h)  # [T, h]
        t = xf.shape[0]
        logits = self.router(xf).float()  # [T, E]
        probs = torch.softmax(logits, dim=-1)
        topv, topi = probs.topk(cfg.top_k, dim=-1)  # [T, K]
        topv = (topv / topv.sum(-1, keepdim=True)).to(x.dtype)

        # load-balancing auxiliary loss (Switch-style). Returned through the
        # forward output (not stashed on the module) so it stays connected to
        # autograd under activation checkpointing, where this code runs in a
        # no-grad first pass and a grad-enabled recompute.
        aux = xf.new_zeros(())
        if self.training:
            me = probs.mean(0)
            ce = F.one_hot(topi[:, 0], cfg.n_experts).float().mean(0)
            aux = (cfg.router_aux_loss_coef * cfg.n_experts * (me * ce).sum()).to(xf.dtype)

        # group tokens by expert: one dense GEMM chain per expert
        flat_expert = topi.reshape(-1)  # [T*K]
        flat_tok = (
            torch.arange(t, device=x.device).unsqueeze(1).expand(-1, cfg.top_k).reshape(-1)
        )
        order = torch.argsort(flat_expert, stable=True)
        counts = torch.bincount(flat_expert, minlength=cfg.n_experts).tolist()
        gathered = xf[flat_tok[order]]  # [T*K, h] grouped by expert

        if self.ep is not None:
            grouped_out = self._ep_dispatch(gathered, counts)
            weights = topv.reshape(-1)[order].unsqueeze(1)
            out = torch.zeros_like(xf)
            out.index_add_(0, flat_tok[order], grouped_out * weights)
            return out.reshape(b, s, h), aux
        # The native grouped kernel measured SLOWER than per-expert hipBLASLt
        # at every tested size (tools/moe_gemm_bench.py: 205-687 vs 250-1154
        # TF/s), so the library loop is the default; HYPHA_NATIVE_GROUPED=1
        # routes inference through the native kernel (building block for the
        # round-2 fused MoE path).
        import os

        use_native_grouped = (
            os.environ.get("HYPHA_NATIVE_GROUPED", "0") == "1"
            and not torch.is_grad_enabled() and gathered.is_cuda
            and gathered.dtype == torch.bfloat16 and ops.has_native()
            and cfg.ffn_hidden % 128 == 0 and cfg.hidden_size % 128 == 0
        )
        if use_native_grouped:
            # inference path: one native grouped-GEMM kernel per projection
            from hypha_amd import _C

            off = torch.zeros(cfg.n_experts + 1, dtype=torch.int32)
            off[1:] = torch.tensor(counts, dtype=torch.int32).cumsum(0)
            ge = _C.grouped_gemm(gathered.contiguous(), self.w_gate.contiguous(), off)
            ue = _C.grouped_gemm(gathered.contiguous(), self.w_up.contiguous(), off)
            he = ops.swiglu(ge, ue)
            grouped_out = _C.grouped_gemm(he.contiguous(), self.w_down.contiguous(), off)
        else:
            grouped_out = self._expert_ffn(gathered, counts)
        # scatter-add back with routing weights
        weights = topv.reshape(-1)[order].unsqueeze(1)
        out = torch.zeros_like(xf)
        out.index_add_(0, flat_tok[order], grouped_out * weights)
        return out.reshape(b, s, h), aux

    def _expert_ffn(self, xs: torch.Tensor, counts: list[int]) -> torch.Tensor:
        """Per-expert SwiGLU chains over expert-grouped rows (local shard)."""
        out_groups = []
        start = 0
        for e in range(len(counts)):
            n = counts[e]
            if n == 0:
                continue
            xe = xs[start:start + n]
            ge = xe @ self.w_gate[e].t()
            ue = xe @ self.w_up[e].t()
            he = ops.swiglu(ge, ue)
            out_groups.append(he @ self.w_down[e].t())
            start += n
        return torch.cat(out_groups, dim=0) if out_groups else xs[:0]

    def _ep_dispatch(self, gathered: torch.Tensor, counts: list[int]) -> torch.Tensor:
        """Expert-parallel token exchange (all-to-all over xGMI/RCCL):

        1. `gathered` is globally expert-sorted, so each rank's experts form
           one contiguous send chunk;
        2. every rank learns the full [size, E] count matrix (all_gather) to
           size the exchange and regroup received rows expert-major;
        3. local expert GEMMs, then the transposed all-to-all returns rows
           in the original `gathered` order for the weighted combine."""
        import torch.distributed as dist

        rank, size, group = self.ep
        cfg = self.cfg
        e_local = cfg.n_experts // size
        dev = gathered.device

        counts_t = torch.tensor(counts, dtype=torch.int64, device=dev)
        all_counts = [torch.empty_like(counts_t) for _ in range(size)]
        dist.all_gather(all_counts, counts_t, group=group)
        L = torch.stack(all_counts).cpu()  # [size, E]

        send_splits = [int(L[rank, r * e_local:(r + 1) * e_local].sum())
                       for r in range(size)]
        my = slice(rank * e_local, (rank + 1) * e_local)
        recv_splits = [int(L[src, my].sum()) for src in range(size)]

        recv = _AllToAll.apply(gathered, recv_splits, send_splits, group)

        # regroup src-major -> expert-major for the GEMM chains
        seg = []  # (src, e) segment order as received
        off = 0
        pos = {}
        for src in range(size):
            for e in range(e_local):
                n = int(L[src, rank * e_local + e])
                pos[(src, e)] = (off, n)
                off += n
        perm = []
        local_counts = []
        for e in range(e_local):
            c = 0
            for src in range(size):
                o, n = pos[(src, e)]
                if n:
                    perm.append(torch.arange(o, o + n, device=dev))
                c += n
            local_counts.append(c)
        perm = torch.cat(perm) if perm else torch.empty(0, dtype=torch.long, device=dev)
        inv = torch.empty_like(perm)
        inv[perm] = torch.arange(perm.numel(), device=dev)

        ys = self._expert_ffn(recv[perm], local_counts)
        back = _AllToAll.apply(ys[inv], send_splits, recv_splits, group)
        return back


def shard_experts_(model: nn.Module, rank: int, size: int, group=None) -> int:
    """EXPERT PARALLELISM (opt-in): keep only this rank's n_experts/size
    expert slices in every MoEMLP; tokens reach remote experts by RCCL
    all-to-all each step. Sharded parameters are tagged `_ep_local` so the
    DiLoCo engines exclude them from the outer all-reduce (each expert is a
    singleton that already saw every rank's tokens) and from the init
    broadcast. Returns the number of sharded MoEMLP modules."""
    import torch.distributed as dist

    if size <= 1:
        return 0  # single rank: the dense path IS expert-parallel degree 1
    if group is None:
        group = dist.distributed_c10d._get_default_group()
    n = 0
    for m in model.modules():
        if isinstance(m, MoEMLP):
            cfg = m.cfg
            assert cfg.n_experts % size == 0, "experts must divide EP size"
            sl = slice(rank * (cfg.n_experts // size),
                       (rank + 1) * (cfg.n_experts // size))
            for name in ("w_gate", "w_up", "w_down"):
                p = getattr(m, name)
                new = nn.Parameter(p.data[sl].clone())
                new._ep_local = True
                setattr(m, name, new)
            m.ep = (rank, size, group)
            n += 1
    return n


class MoEBlock(nn.Module):
    def __init__(self, cfg: MoEConfig):
        super().__init__()
        self.attn_norm = RMSNorm(cfg.hidden_size, cfg.norm_eps)
        self.attn = Attention(cfg)
        self.mlp_norm = RMSNorm(cfg.hidden_size, cfg.norm_eps)
        self.mlp = MoEMLP(cfg)

    def forward(self, x, cos, sin, cache=None, pos: int = 0):
        attn_out = self.attn(self.attn_norm(x), cos, sin, cache=cache, pos=pos)
        x, n2 = ops.add_rmsnorm(x, attn_out, self.mlp_norm.weight,
                                self.mlp_norm.eps)
        mlp_out, aux = self.mlp(n2)
        x = x + mlp_out
        return x, aux

    def forward_pair(self, res, delta, cos, sin):
        """Training fast path: deferred residual adds fused into the norm
        kernels (see llama.Block.forward_pair)."""
        res, n1 = ops.add_rmsnorm(res, delta, self.attn_norm.weight,
                                  self.attn_norm.eps)
        attn_out = self.attn(n1, cos, sin)
        res, n2 = ops.add_rmsnorm(res, attn_out, self.mlp_norm.weight,
                                  self.mlp_norm.eps)
        mlp_out, aux = self.mlp(n2)
        return res, mlp_out, aux


class MoEForCausalLM(nn.Module):
    def __init__(self, cfg: MoEConfig):
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.blocks = nn.ModuleList(MoEBlock(cfg) for _ in range(cfg.n_layers))
        self.norm = RMSNorm(cfg.hidden_size, cfg.norm_eps)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        cos, sin = ops.reference.rope_cos_sin(cfg.max_seq_len, cfg.head_dim, cfg.rope_base)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        nn.init.normal_(self.embed.weight, std=cfg.init_std)
        nn.init.normal_(self.lm_head.weight, std=cfg.init_std)

    def forward(self, input_ids, labels=None):
        import torch.utils.checkpoint

        x = self.embed(input_ids)
        aux_total = None
        delta = torch.zeros_like(x)
        for blk in self.blocks:
            if self.cfg.gradient_checkpointing and self.training:
                x, delta, aux = torch.utils.checkpoint.checkpoint(
                    blk.forward_pair, x, delta, self.rope_cos, self.rope_sin,
                    use_reentrant=False)
            else:
                x, delta, aux = blk.forward_pair(x, delta, self.rope_cos,
                                                 self.rope_sin)
            aux_total = aux if aux_total is None else aux_total + aux
        _, x = ops.add_rmsnorm(x, delta, self.norm.weight, self.norm.eps)
        logits = self.lm_head(x)
        if labels is None:
            return logits
        loss = ops.cross_entropy_loss(logits[:, :-1], labels[:, 1:])
        if self.training and aux_total is not None:
            loss = loss + aux_total.float()
        return loss

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int = 32,
                 temperature: float = 0.0, top_k: int = 0,
                 seed: int | None = None) -> torch.Tensor:
        """KV-cache decoding (prefill once, then one token per step via the
        flash-decoding kernel; same sampling surface as the Llama family so
        inference jobs can run any registry model)."""
        from .kv_cache import KVCache

        self.eval()
        b, s = input_ids.shape
        dtype = self.embed.weight.dtype
        caches = [KVCache(b, s + max_new_tokens, self.cfg.n_kv_heads,
                          self.cfg.head_dim, input_ids.device, dtype=dtype)
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
                x, _ = blk(x, self.rope_cos, self.rope_sin, cache=cache, pos=pos)
            x = self.norm(x[:, -1:])
            logits = self.lm_head(x)[:, 0]
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


def build_model(name: str, **overrides) -> MoEForCausalLM:
    cfg = PRESETS[name]
    if overrides:
        from dataclasses import replace

        cfg = replace(cfg, **overrides)
    return MoEForCausalLM(cfg)
