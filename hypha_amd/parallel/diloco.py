"""DiLoCo engine: flat-bucket parameter management + fused inner-AdamW +
outer-Nesterov synchronization over RCCL.

Device-resident redesign of the reference's file-based DiLoCo loop
(/root/reference/executors/accelerate/src/hypha/accelerate_executor/training.py
and crates/worker/src/executor/parameter_server.rs): where the reference
round-trips every sync through safetensors files and libp2p streams, here
theta_0, AdamW state, the pseudo-gradient and the outer momentum live in flat
GPU buffers sized for 288 GB HBM3E; the outer sync is a bucketed RCCL
all-reduce followed by an identical (replicated) fused Nesterov step on every
rank — no parameter-server star, no broadcast needed.

Sign convention (must match the reference, utils.py:118-123): the
pseudo-gradient is delta = theta_t - theta_0 (the negative of a gradient);
the outer step ADDS lr*(mu*m + delta).
"""

from __future__ import annotations

from dataclasses import dataclass, field

import torch
import torch.nn as nn

from hypha_amd import ops
from .comm import Comm


@dataclass
class InnerOptConfig:
    lr: float = 4e-4
    beta1: float = 0.9
    beta2: float = 0.95
    eps: float = 1e-8
    weight_decay: float = 0.1
    warmup_steps: int = 10
    schedule: str = "cosine"  # constant | cosine | linear | wsd
    total_steps: int = 10000
    min_lr_frac: float = 0.1
    state_bits: int = 32  # 32 = fp32 m/v; 8 = blockwise-uint8 m/v (config 5)


@dataclass
class OuterOptConfig:
    lr: float = 0.7
    momentum: float = 0.9


@dataclass
class DiLoCoConfig:
    h: int = 100  # inner steps between outer syncs
    inner: InnerOptConfig = field(default_factory=InnerOptConfig)
    outer: OuterOptConfig = field(default_factory=OuterOptConfig)
    comm_dtype: torch.dtype = torch.bfloat16
    grad_clip: float = 1.0


def lr_at(cfg: InnerOptConfig, step: int) -> float:
    """LR schedules mirroring the reference's constant/cosine/linear/wsd set
    (executors/accelerate/.../utils.py:94-106)."""
    import math

    if step < cfg.warmup_steps:
        return cfg.lr * (step + 1) / max(1, cfg.warmup_steps)
    if cfg.schedule == "constant":
        return cfg.lr
    t = min(1.0, (step - cfg.warmup_steps) / max(1, cfg.total_steps - cfg.warmup_steps))
    if cfg.schedule == "cosine":
        return cfg.lr * (cfg.min_lr_frac + (1 - cfg.min_lr_frac) * 0.5 * (1 + math.cos(math.pi * t)))
    if cfg.schedule == "linear":
        return cfg.lr * (1 - (1 - cfg.min_lr_frac) * t)
    if cfg.schedule == "wsd":  # warmup-stable-decay: decay in last 10%
        if t < 0.9:
            return cfg.lr
        return cfg.lr * (1 - (1 - cfg.min_lr_frac) * (t - 0.9) / 0.1)
    raise ValueError(cfg.schedule)


class FlatParams:
    """All trainable params flattened into one contiguous working buffer,
    with fp32 master/optimizer/outer state as matching flat buffers."""

    QBLOCK = 2048  # must match the adamw8 kernel's quant block

    def __init__(self, model: nn.Module, device: torch.device, work_dtype: torch.dtype,
                 state_bits: int = 32):
        # expert-parallel shards (`_ep_local`, models/moe.py shard_experts_)
        # are placed LAST: the outer all-reduce and the init broadcast cover
        # only the shared prefix [0, sync_numel) — each expert shard is a
        # singleton owned by one rank, not a replica to average.
        params = [p for p in model.parameters() if p.requires_grad]
        shared = [p for p in params if not getattr(p, "_ep_local", False)]
        ep = [p for p in params if getattr(p, "_ep_local", False)]
        self.params = shared + ep
        self.sync_numel = sum(p.numel() for p in shared)
        self.numel = sum(p.numel() for p in self.params)
        self.device = device
        self.work_dtype = work_dtype

        self.flat = torch.empty(self.numel, dtype=work_dtype, device=device)
        self.flat_grad = torch.zeros(self.numel, dtype=work_dtype, device=device)
        offset = 0
        for p in self.params:
            n = p.numel()
            self.flat[offset : offset + n].copy_(p.data.reshape(-1).to(work_dtype))
            p.data = self.flat[offset : offset + n].view(p.shape)
            p.grad = self.flat_grad[offset : offset + n].view(p.shape)
            offset += n

        self.master = self.flat.float()  # fp32 master weights
        self.state_bits = state_bits
        if state_bits == 8 and device.type == "cuda":
            nblocks = (self.numel + self.QBLOCK - 1) // self.QBLOCK
            self.m8 = torch.full((self.numel,), 127, dtype=torch.uint8, device=device)
            self.v8 = torch.zeros(self.numel, dtype=torch.uint8, device=device)
            self.m_scale = torch.full((nblocks,), 1e-12, device=device)
            self.v_scale = torch.full((nblocks,), 1e-12, device=device)
            self.exp_avg = self.exp_avg_sq = None
        else:
            self.state_bits = 32
            self.exp_avg = torch.zeros_like(self.master)
            self.exp_avg_sq = torch.zeros_like(self.master)
        self.theta0 = self.master.clone()  # global weights at round start
        self.outer_momentum = torch.zeros_like(self.master)

    def zero_grad(self) -> None:
        self.flat_grad.zero_()

    def grad_norm(self) -> torch.Tensor:
        # fp32 accumulation without materializing an fp32 copy of the grads
        if self.flat_grad.is_cuda and self.flat_grad.dtype == torch.bfloat16:
            from hypha_amd import _C

            return _C.grad_norm_sq(self.flat_grad).sqrt_()
        return torch.linalg.vector_norm(self.flat_grad, dtype=torch.float32)


class DiLoCoWorker:
    """One worker peer = one GPU = one process. Runs H fused inner-AdamW steps,
    then an outer RCCL all-reduce + fused Nesterov."""

    def __init__(
        self,
        model: nn.Module,
        cfg: DiLoCoConfig,
        comm: Comm | None = None,
        device: torch.device | None = None,
    ):
        self.model = model
        self.cfg = cfg
        self.comm = comm if comm is not None else Comm()
        if device is None:
            device = (
                torch.device("cuda", self.comm.local_rank)
                if torch.cuda.is_available()
                else torch.device("cpu")
            )
        self.device = device
        work_dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
        model.to(device=device, dtype=work_dtype)
        # buffers (rope tables) stay fp32 for precision
        for buf in model.buffers():
            if buf.dtype in (torch.bfloat16, torch.float16):
                buf.data = buf.data.float()
        self.fp = FlatParams(model, device, work_dtype, state_bits=cfg.inner.state_bits)
        # make every rank start from identical weights (rank0's init wins);
        # expert-parallel shards (the tail past sync_numel) stay rank-local
        if self.comm.is_distributed:
            self.comm.broadcast_flat(self.fp.master[: self.fp.sync_numel], src=0)
            self.fp.flat.copy_(self.fp.master)
            self.fp.theta0.copy_(self.fp.master)
        self.inner_step_count = 0  # global inner step counter (for LR/bias corr)
        self.round = 0  # outer rounds completed
        self.steps_in_round = 0
        # staging buffer for chunked extract+allreduce (bounded memory)
        self._comm_chunk = min(self.fp.numel, 64 * 1024 * 1024)
        self._delta_buf = torch.empty(
            self._comm_chunk, dtype=cfg.comm_dtype, device=device
        )
        self.outer_sync_payload_bytes = 0
        self.last_loss: float = float("nan")

    # -- inner ------------------------------------------------------------

    def train_step(self, input_ids: torch.Tensor, labels: torch.Tensor) -> float:
        """One inner step: forward, backward, fused AdamW. Returns loss."""
        from hypha_amd.ops.fp8 import fp8_step

        fp8_step()  # new fp8 epoch: weight casts refresh once per step
        self.model.train()
        input_ids = input_ids.to(self.device, non_blocking=True)
        labels = labels.to(self.device, non_blocking=True)
        loss = self.model(input_ids, labels=labels)
        loss.backward()
        self.inner_opt_step()
        self.last_loss = float(loss.detach().float().cpu())
        return self.last_loss

    def inner_opt_step(self) -> None:
        """Fused AdamW over the flat buffers (one kernel launch on GPU)."""
        self.inner_step_count += 1
        self.steps_in_round += 1
        c = self.cfg.inner
        if self.cfg.grad_clip > 0:
            gnorm = self.fp.grad_norm()
            # clamp to 1 and always scale: no host sync, one fused mul
            clip_coef = (self.cfg.grad_clip / (gnorm + 1e-6)).clamp_(max=1.0)
            self.fp.flat_grad.mul_(clip_coef.to(self.fp.flat_grad.dtype))
        if self.fp.state_bits == 8:
            from hypha_amd import _C

            _C.adamw8_step_(
                self.fp.master, self.fp.flat, self.fp.flat_grad,
                self.fp.m8, self.fp.v8, self.fp.m_scale, self.fp.v_scale,
                lr_at(c, self.inner_step_count - 1), c.beta1, c.beta2, c.eps,
                c.weight_decay, self.inner_step_count,
            )
        else:
            ops.fused_adamw(
                self.fp.master,
                self.fp.flat,
                self.fp.flat_grad,
                self.fp.exp_avg,
                self.fp.exp_avg_sq,
                lr=lr_at(c, self.inner_step_count - 1),
                beta1=c.beta1,
                beta2=c.beta2,
                eps=c.eps,
                weight_decay=c.weight_decay,
                step=self.inner_step_count,
            )
        self.fp.zero_grad()

    # -- outer ------------------------------------------------------------

    def maybe_outer_sync(self) -> bool:
        if self.steps_in_round >= self.cfg.h:
            self.outer_sync()
            return True
        return False

    def outer_sync(self) -> None:
        """delta = master - theta0; all-reduce(mean); fused Nesterov on theta0;
        master/params <- new global weights. Chunked so the comm staging buffer
        is O(64MB) regardless of model size."""
        fp = self.fp
        n = fp.numel
        n_sync = fp.sync_numel  # expert-parallel tail is not averaged
        chunk = self._comm_chunk
        for start in range(0, n, chunk):
            m = min(chunk, n - start)
            ms = max(0, min(m, n_sync - start))  # shared portion of chunk
            if ms > 0:
                d = self._delta_buf[:ms]
                ops.interface.extract_delta(
                    fp.master[start : start + ms], fp.theta0[start : start + ms], d
                )
                self.comm.all_reduce_mean_flat(d)
                ops.fused_nesterov(
                    fp.theta0[start : start + ms],
                    d,
                    fp.outer_momentum[start : start + ms],
                    lr=self.cfg.outer.lr,
                    mu=self.cfg.outer.momentum,
                )
            if ms < m:
                # expert-parallel tail: singletons, purely inner-trained —
                # the round's weights pass through (no averaging, no outer
                # momentum re-amplifying an un-averaged delta)
                fp.theta0[start + ms : start + m].copy_(
                    fp.master[start + ms : start + m]
                )
        fp.master.copy_(fp.theta0)
        fp.flat.copy_(fp.master)
        self.outer_sync_payload_bytes += n_sync * self._delta_buf.element_size()
        self.round += 1
        self.steps_in_round = 0

    # -- reporting ---------------------------------------------------------

    def comm_stats(self) -> dict:
        payload = self.outer_sync_payload_bytes
        return {
            "outer_rounds": self.round,
            "outer_sync_payload_bytes": payload,
            "outer_sync_wire_bytes_per_rank": self.comm.wire_bytes_per_rank(payload),
            "h": self.cfg.h,
            "model_numel": self.fp.numel,
        }
