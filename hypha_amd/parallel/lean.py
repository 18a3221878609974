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
                 device: torch.device | None = None):
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

        # flatten bf16 params (the single weight copy)
        self.params = [p for p in model.parameters() if p.requires_grad]
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
            self.comm.broadcast_flat(self.flat, src=0)

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

        chunk = min(self.numel, 64 * 1024 * 1024)
        self._chunk = chunk
        self._stage_t0 = torch.empty(chunk, dtype=torch.bfloat16, device=self.device)
        self._stage_m = torch.empty(chunk, dtype=torch.bfloat16, device=self.device)
        self._stage_d = torch.empty(chunk, dtype=torch.bfloat16, device=self.device)

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
        seed = (self.round + 1) * 40503 % (2**31)
        for start in range(0, n, self._chunk):
            m = min(self._chunk, n - start)
            t0 = self._stage_t0[:m]
            mo = self._stage_m[:m]
            d = self._stage_d[:m]
            t0.copy_(self.theta0_host[start:start + m], non_blocking=False)
            mo.copy_(self.outer_m_host[start:start + m], non_blocking=False)
            self._C.extract_delta_bf16(self.flat[start:start + m], t0, d)
            self.comm.all_reduce_mean_flat(d)
            self._C.nesterov_bf16_(t0, d, mo, self.cfg.outer.lr,
                                   self.cfg.outer.momentum, seed)
            self.theta0_host[start:start + m].copy_(t0)
            self.outer_m_host[start:start + m].copy_(mo)
            self.flat[start:start + m].copy_(t0)
        self.outer_sync_payload_bytes += n * 2
        self.round += 1
        self.steps_in_round = 0

    def comm_stats(self) -> dict:
        payload = self.outer_sync_payload_bytes
        return {
            "outer_rounds": self.round,
            "outer_sync_payload_bytes": payload,
            "outer_sync_wire_bytes_per_rank": self.comm.wire_bytes_per_rank(payload),
            "h": self.cfg.h,
            "model_numel": self.numel,
            "memory_mode": "lean",
        }
