"""Lean-memory DiLoCo engine (BASELINE configs 4/5): trains models whose
fp32 optimizer state cannot fit in 288 GB HBM (Mixtral-8x7B 46.7B params,
Llama-3-70B).

Device-memory layout per worker:
  * bf16 parameters are the ONLY weight copy (no fp32 master): the fused
    AdamW applies updates with stochastic rounding (unbiased, so small
    updates accumulate in expectation instead of vanishing to RNE).
  * AdamW m/v are blockwise-uint8 (sqrt-domain v), 2 B/param total.
  * gradients are RELEASED during backward: a post-accumulate hook applies
    the fused optimizer to each parameter the moment its grad is ready and
    frees it — peak grad memory is a few parameters, not the full model.
  * theta0 (round-start global weights) and the outer Nesterov momentum are
    HOST-resident bf16, streamed through a 64 MB staging window during the
    outer sync (every H steps; ~PCIe-seconds amortized to ~nothing).

Device bytes ~ 2P (params) + 2P (m8+v8) + staging. Mixtral-8x7B: ~187 GB.
Trade-offs vs the fp32-state engine: no global grad-norm clipping (grads
never coexist) and stochastically-rounded weight updates.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .comm import Comm
from .diloco import DiLoCoConfig, lr_at

QBLOCK = 2048


class LeanDiLoCoWorker:
    def __init__(self, model: nn.Module, cfg: DiLoCoConfig, comm: Comm | None = None,
                 device: torch.device | None = None, fp8_weights: bool = False):
        from hypha_amd import ops

        assert torch.cuda.is_available(), "lean mode needs a GPU"
        ops.native_available_or_raise()
        from hypha_amd import _C

        self._C = _C
        self.model = model
        self.cfg = cfg
        self.comm = comm if comm is not None else Comm()
        self.device = device or torch.device("cuda", self.comm.local_rank)
        model.to(device=self.device, dtype=torch.bfloat16)
        for buf in model.buffers():
            if buf.dtype in (torch.bfloat16, torch.float16):
                buf.data = buf.data.float()

        # fp8 WEIGHT STORAGE (BASELINE config 5): every Fp8Linear's weight
        # moves into e4m3 + per-2048-block scales (1 B/param instead of 2);
        # their dw is applied by the fused fp8 AdamW inside backward. The
        # remaining (bf16) params follow the original lean path below.
        self.fp8_weights = fp8_weights
        self._fp8_mods = []
        if fp8_weights:
            from hypha_amd.ops.fp8 import Fp8Linear, convert_linears_to_fp8

            convert_linears_to_fp8(model)
            for m in model.modules():
                if isinstance(m, Fp8Linear) and m.weight is not None:
                    assert m.weight.numel() % QBLOCK == 0, (
                        "fp8 weight storage needs QBLOCK-aligned shapes")
                    self._fp8_mods.append(m)

        # flatten bf16 params (the single weight copy). Expert-parallel
        # shards (`_ep_local`, models/moe.py shard_experts_) go LAST: the
        # outer all-reduce and init broadcast cover only [0, sync_numel) —
        # expert singletons are never averaged (see diloco.FlatParams).
        fp8_param_ids = {id(m.weight) for m in self._fp8_mods}
        all_params = [p for p in model.parameters()
                      if p.requires_grad and id(p) not in fp8_param_ids]
        shared = [p for p in all_params if not getattr(p, "_ep_local", False)]
        ep_tail = [p for p in all_params if getattr(p, "_ep_local", False)]
        self.params = shared + ep_tail
        self.sync_numel = sum(p.numel() for p in shared)
        self.numel = sum(p.numel() for p in self.params)
        self.flat = torch.empty(self.numel, dtype=torch.bfloat16, device=self.device)
        offset = 0
        self._offsets = []
        for p in self.params:
            n = p.numel()
            self.flat[offset:offset + n].copy_(p.data.reshape(-1))
            p.data = self.flat[offset:offset + n].view(p.shape)
            self._offsets.append(offset)
            offset += n

        if self.comm.is_distributed:
            self.comm.broadcast_flat(self.flat[: self.sync_numel], src=0)

        # per-param quant-state slices, padded to the kernel's 2048 block
        self._state_off = []
        total = 0
        for p in self.params:
            self._state_off.append(total)
            total += (p.numel() + QBLOCK - 1) // QBLOCK * QBLOCK
        self.m8 = torch.full((total,), 127, dtype=torch.uint8, device=self.device)
        self.v8 = torch.zeros(total, dtype=torch.uint8, device=self.device)
        nblocks = total // QBLOCK
        self.m_scale = torch.full((nblocks,), 1e-12, device=self.device)
        self.v_scale = torch.full((nblocks,), 1e-12, device=self.device)

        # ---- fp8 weight-storage group (config 5) ----
        self.fp8_numel = sum(m.weight.numel() for m in self._fp8_mods)
        if self.fp8_numel:
            n8 = self.fp8_numel
            self.flat_w8 = torch.empty(n8, dtype=torch.uint8, device=self.device)
            self.flat_wscale = torch.empty(n8 // QBLOCK, dtype=torch.float32,
                                           device=self.device)
            # quantize + FREE the bf16 weights module by module BEFORE
            # allocating optimizer state: peak device memory stays
            # bf16(live) + w8, never bf16 + w8 + m8/v8 (70B would OOM)
            off = 0
            self._fp8_off = []
            for m in self._fp8_mods:
                n = m.weight.numel()
                shape = m.weight.shape
                _C.fp8_requant_(m.weight.data.reshape(-1).contiguous(),
                                self.flat_w8[off:off + n],
                                self.flat_wscale[off // QBLOCK:(off + n) // QBLOCK],
                                40503)
                m.weight = None  # fp8 is now the only weight copy
                m.lean_w8 = self.flat_w8[off:off + n].view(shape)
                m.lean_wscale = self.flat_wscale[off // QBLOCK:(off + n) // QBLOCK]
                m.lean_opt_hook = self._make_fp8_hook(off, n)
                self._fp8_off.append(off)
                off += n
            torch.cuda.empty_cache()
            self.m8_f = torch.full((n8,), 127, dtype=torch.uint8, device=self.device)
            self.v8_f = torch.zeros(n8, dtype=torch.uint8, device=self.device)
            self.ms_f = torch.full((n8 // QBLOCK,), 1e-12, device=self.device)
            self.vs_f = torch.full((n8 // QBLOCK,), 1e-12, device=self.device)
            if self.comm.is_distributed:
                self.comm.broadcast_flat(self.flat_w8, src=0)
                self.comm.broadcast_flat(self.flat_wscale, src=0)
            # host-resident outer state for the fp8 group, built chunk-wise
            # (a full-size device dequant buffer would be another 2P bytes)
            self.theta0_fp8_host = torch.empty(n8, dtype=torch.bfloat16)
            chunk0 = min(n8, 64 * 1024 * 1024)
            zero = torch.zeros(chunk0, dtype=torch.bfloat16, device=self.device)
            dq = torch.empty(chunk0, dtype=torch.bfloat16, device=self.device)
            for s in range(0, n8, chunk0):
                e = min(s + chunk0, n8)
                _C.fp8_extract_delta(self.flat_w8[s:e],
                                     self.flat_wscale[s // QBLOCK:(e + QBLOCK - 1) // QBLOCK],
                                     zero[:e - s], dq[:e - s])
                self.theta0_fp8_host[s:e].copy_(dq[:e - s])
            self.outer_m_fp8_host = torch.zeros_like(self.theta0_fp8_host)
            del dq, zero

        # host-resident outer state (bf16)
        self.theta0_host = self.flat.detach().cpu().clone()
        self.outer_m_host = torch.zeros_like(self.theta0_host)

        # grad-release hooks: optimizer applies per-param during backward
        self._cur_lr = cfg.inner.lr
        self.inner_step_count = 0
        self.round = 0
        self.steps_in_round = 0
        self.outer_sync_payload_bytes = 0
        self.last_loss = float("nan")
        for idx, p in enumerate(self.params):
            p.register_post_accumulate_grad_hook(self._make_hook(idx))

        chunk = min(max(self.numel, self.fp8_numel), 64 * 1024 * 1024)
        self._chunk = chunk
        self._stage_t0 = torch.empty(chunk, dtype=torch.bfloat16, device=self.device)
        self._stage_m = torch.empty(chunk, dtype=torch.bfloat16, device=self.device)
        self._stage_d = torch.empty(chunk, dtype=torch.bfloat16, device=self.device)

    def _make_fp8_hook(self, off: int, n: int):
        c = self.cfg.inner
        b0, b1 = off // QBLOCK, (off + n) // QBLOCK

        def hook(dw: torch.Tensor):
            self._C.adamw8_fp8_lean_(
                self.flat_w8[off:off + n], self.flat_wscale[b0:b1],
                dw.reshape(-1), self.m8_f[off:off + n], self.v8_f[off:off + n],
                self.ms_f[b0:b1], self.vs_f[b0:b1],
                self._cur_lr, c.beta1, c.beta2, c.eps, c.weight_decay,
                self.inner_step_count, self.inner_step_count * 2654435761 % (2**31),
            )

        return hook

    def _make_hook(self, idx: int):
        c = self.cfg.inner

        def hook(p: torch.Tensor):
            n = p.numel()
            so = self._state_off[idx]
            npad = (n + QBLOCK - 1) // QBLOCK * QBLOCK
            self._C.adamw8_lean_(
                p.data.view(-1), p.grad.view(-1),
                self.m8[so:so + npad], self.v8[so:so + npad],
                self.m_scale[so // QBLOCK:(so + npad) // QBLOCK],
                self.v_scale[so // QBLOCK:(so + npad) // QBLOCK],
                self._cur_lr, c.beta1, c.beta2, c.eps, c.weight_decay,
                self.inner_step_count, self.inner_step_count * 2654435761 % (2**31),
            )
            p.grad = None  # release immediately

        return hook

    def train_step(self, input_ids: torch.Tensor, labels: torch.Tensor) -> float:
        from hypha_amd.ops.fp8 import fp8_step

        fp8_step()  # new fp8 epoch: weight casts refresh once per step
        self.model.train()
        self.inner_step_count += 1
        self.steps_in_round += 1
        self._cur_lr = lr_at(self.cfg.inner, self.inner_step_count - 1)
        input_ids = input_ids.to(self.device, non_blocking=True)
        labels = labels.to(self.device, non_blocking=True)
        loss = self.model(input_ids, labels=labels)
        loss.backward()  # hooks run the optimizer and free each grad
        self.last_loss = float(loss.detach().float().cpu())
        return self.last_loss

    def maybe_outer_sync(self) -> bool:
        if self.steps_in_round >= self.cfg.h:
            self.outer_sync()
            return True
        return False

    @torch.no_grad()
    def outer_sync(self) -> None:
        """Chunked: stream host theta0/momentum through the staging window,
        delta = theta_t - theta0, all-reduce, Nesterov, write back."""
        n = self.numel
        n_sync = self.sync_numel  # EP expert tail is rank-local
        seed = (self.round + 1) * 40503 % (2**31)
        for start in range(0, n, self._chunk):
            m = min(self._chunk, n - start)
            ms = max(0, min(m, n_sync - start))  # shared portion
            if ms > 0:
                t0 = self._stage_t0[:ms]
                mo = self._stage_m[:ms]
                d = self._stage_d[:ms]
                t0.copy_(self.theta0_host[start:start + ms], non_blocking=False)
                mo.copy_(self.outer_m_host[start:start + ms], non_blocking=False)
                self._C.extract_delta_bf16(self.flat[start:start + ms], t0, d)
                self.comm.all_reduce_mean_flat(d)
                self._C.nesterov_bf16_(t0, d, mo, self.cfg.outer.lr,
                                       self.cfg.outer.momentum, seed)
                self.theta0_host[start:start + ms].copy_(t0)
                self.outer_m_host[start:start + ms].copy_(mo)
                self.flat[start:start + ms].copy_(t0)
            if ms < m:
                # expert-parallel tail: purely inner-trained singletons pass
                # through the round (no averaging, no outer momentum)
                self.theta0_host[start + ms:start + m].copy_(
                    self.flat[start + ms:start + m])
        # fp8 weight-storage group: dequant-delta, all-reduce, Nesterov on
        # the bf16 global weights, SR-requant back into storage
        if self.fp8_numel:
            n8 = self.fp8_numel
            rseed = (self.round + 1) * 48611 % (2**31)
            for start in range(0, n8, self._chunk):
                m = min(self._chunk, n8 - start)
                t0 = self._stage_t0[:m]
                mo = self._stage_m[:m]
                d = self._stage_d[:m]
                t0.copy_(self.theta0_fp8_host[start:start + m], non_blocking=False)
                mo.copy_(self.outer_m_fp8_host[start:start + m], non_blocking=False)
                b0 = start // QBLOCK
                b1 = (start + m + QBLOCK - 1) // QBLOCK
                self._C.fp8_extract_delta(self.flat_w8[start:start + m],
                                          self.flat_wscale[b0:b1], t0, d)
                self.comm.all_reduce_mean_flat(d)
                self._C.nesterov_bf16_(t0, d, mo, self.cfg.outer.lr,
                                       self.cfg.outer.momentum, rseed)
                self.theta0_fp8_host[start:start + m].copy_(t0)
                self.outer_m_fp8_host[start:start + m].copy_(mo)
                self._C.fp8_requant_(t0, self.flat_w8[start:start + m],
                                     self.flat_wscale[b0:b1], rseed ^ 0x5bd1e995)
            self.outer_sync_payload_bytes += n8 * 2

        self.outer_sync_payload_bytes += n_sync * 2
        self.round += 1
        self.steps_in_round = 0

    def comm_stats(self) -> dict:
        payload = self.outer_sync_payload_bytes
        return {
            "outer_rounds": self.round,
            "outer_sync_payload_bytes": payload,
            "outer_sync_wire_bytes_per_rank": self.comm.wire_bytes_per_rank(payload),
            "h": self.cfg.h,
            "model_numel": self.numel + self.fp8_numel,
            "memory_mode": "lean-fp8" if self.fp8_numel else "lean",
        }
