"""hipGraph-captured greedy decoding for the Llama family.

Eager single-token decode is LAUNCH-bound (measured 6.6 ms/step at b8 for
Llama-3-8B: ~300 kernels x ~4-6 us of launch+epilogue overhead each, while
the GEMMs themselves need ~2 ms of HBM time). The whole per-token step —
embed, per-layer norm/projections/RoPE/cache-append/decode-attention/MLP,
final norm, logits, argmax, bookkeeping — is captured ONCE as a hipGraph
and replayed per token with zero host round-trips:

* the cache length lives in a device int32 read by the decode kernel at
  kernel time (ops/hip/decode.hip attn_decode_graph), so one capture
  serves every step as the cache grows;
* RoPE rows are gathered with a device position index; the cache append is
  index_copy_ with the same index; the sampled token is written back into
  the graph's own input buffer, so replays chain with no synchronization.

Greedy only — temperature sampling stays on the eager path.
"""

from __future__ import annotations

import torch

from hypha_amd import ops
from hypha_amd.models.kv_cache import KVCache


class GraphedDecoder:
    def __init__(self, model, batch: int, prompt_len: int, max_new: int,
                 kv_quant: str | None = None):
        from hypha_amd import _C

        self._C = _C
        self.model = model
        cfg = model.cfg
        dev = next(model.parameters()).device
        self.dev = dev
        self.batch = batch
        self.max_new = max_new
        self.kv_quant = kv_quant
        t_alloc = prompt_len + max_new
        self.caches = [KVCache(batch, t_alloc, cfg.n_kv_heads, cfg.head_dim,
                               dev, dtype=next(model.parameters()).dtype,
                               quant=kv_quant)
                       for _ in model.blocks]
        self.tok = torch.zeros(batch, 1, dtype=torch.long, device=dev)
        self.pos = torch.zeros(1, dtype=torch.long, device=dev)
        self.t32 = torch.zeros(1, dtype=torch.int32, device=dev)
        self.step = torch.zeros(1, dtype=torch.long, device=dev)
        self.out = torch.zeros(batch, max_new, dtype=torch.long, device=dev)
        self.graph = None

    def _step(self) -> None:
        """One decode step over static device state (graph-capturable).
        GEMMs run 2-D through hipBLASLt (the hand-written skinny-M GEMV,
        ops/hip/skinny_gemm.hip, measured 0.4 TB/s vs the library's 1.8-5.9
        on these shapes — shuffle-reduction issue-bound; kept as an opt-in
        negative result like grouped_gemm)."""
        m = self.model
        cfg = m.cfg
        b = self.batch
        lin = lambda t, wt: t @ wt.t()  # noqa: E731
        x = m.embed(self.tok)
        cos = m.rope_cos.index_select(0, self.pos)
        sin = m.rope_sin.index_select(0, self.pos)
        for blk, cache in zip(m.blocks, self.caches):
            n1 = blk.attn_norm(x)[:, 0]  # [b, h]
            a = blk.attn
            q = lin(n1, a.wq.weight).view(b, 1, cfg.n_heads, cfg.head_dim)
            k = lin(n1, a.wk.weight).view(b, 1, cfg.n_kv_heads, cfg.head_dim)
            v = lin(n1, a.wv.weight).view(b, 1, cfg.n_kv_heads, cfg.head_dim)
            q, k = ops.apply_rope_qk(q, k, cos, sin, layout="bshd")
            cache.append_at(k, v, self.pos)
            if self.kv_quant == "fp8":
                o = self._C.attn_decode_fp8_graph(
                    q[:, 0].contiguous(), cache.k, cache.v, cache.k_scale,
                    cache.v_scale, self.t32)
            else:
                o = self._C.attn_decode_graph(q[:, 0].contiguous(), cache.k,
                                              cache.v, self.t32)
            attn_out = lin(o.reshape(b, -1), a.wo.weight).view(b, 1, -1)
            x, n2 = ops.add_rmsnorm(x, attn_out, blk.mlp_norm.weight,
                                    blk.mlp_norm.eps)
            n2f = n2[:, 0]
            h = ops.swiglu(lin(n2f, blk.mlp.w_gate.weight),
                           lin(n2f, blk.mlp.w_up.weight))
            x = x + lin(h, blk.mlp.w_down.weight).view(b, 1, -1)
        x = m.norm(x)
        head_w = m.lm_head.weight if m.lm_head is not None else m.embed.weight
        logits = lin(x[:, 0], head_w)
        nxt = logits.argmax(-1, keepdim=True)  # [b, 1]
        self.step.clamp_(max=self.max_new - 1)
        self.out.index_copy_(1, self.step, nxt)
        self.tok.copy_(nxt)
        self.pos += 1
        self.t32 += 1
        self.step += 1

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int) -> torch.Tensor:
        m = self.model
        b, s = input_ids.shape
        assert b == self.batch and max_new_tokens <= self.max_new
        m.eval()
        for cache in self.caches:  # reuse across generate() calls
            cache.t = 0
        # ---- prefill (eager, MFMA flash kernel) ----
        x = m.embed(input_ids)
        for blk, cache in zip(m.blocks, self.caches):
            x = blk(x, m.rope_cos, m.rope_sin, cache=cache, pos=0)
        xl = m.norm(x[:, -1:])
        logits = (m.lm_head(xl) if m.lm_head is not None
                  else torch.nn.functional.linear(xl, m.embed.weight))
        g0 = logits[:, 0].argmax(-1, keepdim=True)
        self.out[:, 0:1].copy_(g0)
        self.tok.copy_(g0)
        self.pos.fill_(s)
        self.t32.fill_(s + 1)
        self.step.fill_(1)

        n_replay = max_new_tokens - 1
        if n_replay > 0 and self.graph is None:
            if max_new_tokens < 8:  # not worth capturing: run eagerly
                for _ in range(n_replay):
                    self._step()
                return torch.cat([input_ids, self.out[:, :max_new_tokens]], dim=1)
            # 2 warmup steps on a side stream (they are REAL steps), capture
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                self._step()
                self._step()
            torch.cuda.current_stream().wait_stream(side)
            n_replay -= 2
            self.graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph):
                self._step()  # recorded, not executed
        for _ in range(max(0, n_replay)):
            self.graph.replay()
        return torch.cat([input_ids, self.out[:, :max_new_tokens]], dim=1)
