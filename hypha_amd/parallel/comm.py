"""Collective communication over RCCL (xGMI) / gloo.

One process per GPU; backend "nccl" IS RCCL on ROCm. The outer-sync
all-reduce is bucketed: xGMI is 7 point-to-point links per GPU, so ring
collectives are per-link bound — medium-size buckets issued back-to-back let
RCCL pipeline across channels instead of serialising one giant ring pass
(SURVEY.md §2.12). Replaces the reference's libp2p tensor push/pull streams
(C1/C2 in SURVEY.md §2.10; ~1 GB/s ceiling per
/root/reference/rfc/2025-03-25-libp2p_network_stack.md).
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist

DEFAULT_BUCKET_BYTES = 128 * 1024 * 1024


class Comm:
    """Process-group wrapper with byte accounting and single-process fallback."""

    def __init__(self, backend: str | None = None, timeout_s: float = 600.0,
                 rank: int | None = None, world_size: int | None = None,
                 master_addr: str | None = None, master_port: int | None = None):
        """Build from env (torchrun) or from explicit scheduler-assigned
        rendezvous config (rank/world_size/master_addr/master_port in the
        dispatched job spec — the control-plane path, VERDICT r1 item 1,
        replacing the reference's PS star bootstrap
        parameter_server.rs:101-298)."""
        self.rank = rank if rank is not None else int(os.environ.get("RANK", "0"))
        self.world_size = (world_size if world_size is not None
                           else int(os.environ.get("WORLD_SIZE", "1")))
        self.local_rank = int(os.environ.get("LOCAL_RANK", str(self.rank)))
        self.bytes_sent_payload = 0  # payload bytes offered to collectives
        self.syncs = 0
        self.timeout_s = timeout_s
        if self.world_size > 1 and not dist.is_initialized():
            if backend is None:
                backend = "nccl" if torch.cuda.is_available() else "gloo"
            kwargs = {}
            if master_addr is not None or master_port is not None:
                addr = master_addr or "127.0.0.1"
                port = master_port or 29531
                kwargs["init_method"] = f"tcp://{addr}:{port}"
            else:
                os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
                os.environ.setdefault("MASTER_PORT", "29531")
            dist.init_process_group(
                backend=backend,
                rank=self.rank,
                world_size=self.world_size,
                timeout=datetime.timedelta(seconds=timeout_s),
                **kwargs,
            )
        self.backend = dist.get_backend() if dist.is_initialized() else "none"
        if torch.cuda.is_available():
            # the worker daemon pins one GPU per lease via HIP_VISIBLE_DEVICES,
            # so a daemon-spawned executor always sees exactly device 0
            self.local_rank = min(self.local_rank, torch.cuda.device_count() - 1)
            torch.cuda.set_device(self.local_rank)

    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1

    def barrier(self) -> None:
        if self.is_distributed:
            if torch.cuda.is_available() and self.backend == "nccl":
                dist.barrier(device_ids=[self.local_rank])
            else:
                dist.barrier()

    def all_reduce_mean_flat(
        self, flat: torch.Tensor, bucket_bytes: int = DEFAULT_BUCKET_BYTES
    ) -> None:
        """In-place mean all-reduce of a flat tensor, bucketed + async."""
        self.syncs += 1
        self.bytes_sent_payload += flat.numel() * flat.element_size()
        if not self.is_distributed:
            return
        n_per_bucket = max(1, bucket_bytes // flat.element_size())
        handles = []
        for start in range(0, flat.numel(), n_per_bucket):
            chunk = flat.narrow(0, start, min(n_per_bucket, flat.numel() - start))
            handles.append(dist.all_reduce(chunk, op=dist.ReduceOp.SUM, async_op=True))
        for h in handles:
            h.wait()
        flat.div_(self.world_size)

    def broadcast_flat(self, flat: torch.Tensor, src: int = 0) -> None:
        if self.is_distributed:
            dist.broadcast(flat, src=src)

    def all_reduce_scalar(self, value: float, op: str = "max") -> float:
        if not self.is_distributed:
            return value
        device = (
            torch.device("cuda", self.local_rank)
            if (torch.cuda.is_available() and self.backend == "nccl")
            else torch.device("cpu")
        )
        t = torch.tensor([value], dtype=torch.float64, device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX if op == "max" else dist.ReduceOp.SUM)
        return float(t.item())

    def weighted_all_reduce_flat(self, flat: torch.Tensor, weight: float,
                                 bucket_bytes: int = DEFAULT_BUCKET_BYTES) -> float:
        """Sample-weighted aggregate: flat <- sum_i(w_i * flat_i) / sum_i(w_i).

        The reference's PS aggregation weights worker updates by their
        sample counts (the weighting TODO closed on the PS path in round 1);
        this is the same semantics on the RCCL data plane — a rank that
        processed fewer batches this round (heterogeneous FSM schedules,
        mid-round joiners with 0) contributes proportionally less. Returns
        the total weight. With equal weights this equals the plain mean.
        """
        total = self.all_reduce_scalar(float(weight), op="sum")
        if not self.is_distributed:
            return total
        if total <= 0:
            total = float(self.world_size)  # degenerate: plain mean of zeros
        self.syncs += 1
        self.bytes_sent_payload += flat.numel() * flat.element_size()
        flat.mul_(weight / total * self.world_size)  # undo mean's /world
        n_per_bucket = max(1, bucket_bytes // flat.element_size())
        handles = []
        for start in range(0, flat.numel(), n_per_bucket):
            chunk = flat.narrow(0, start, min(n_per_bucket, flat.numel() - start))
            handles.append(dist.all_reduce(chunk, op=dist.ReduceOp.SUM, async_op=True))
        for h in handles:
            h.wait()
        flat.div_(self.world_size)
        return total

    def wire_bytes_per_rank(self, payload_bytes: int) -> int:
        """Ring all-reduce on-wire bytes per rank for a given payload."""
        w = self.world_size
        if w <= 1:
            return 0
        return int(2 * (w - 1) / w * payload_bytes)

    def reform(self, rank: int, world_size: int, master_addr: str = "127.0.0.1",
               master_port: int = 29532, timeout_s: float = 600.0) -> None:
        """Rebuild the process group with a NEW membership (elastic DiLoCo:
        worker kill/rejoin, BASELINE config 3 on the RCCL path).

        RCCL communicators cannot shrink/grow in place; because DiLoCo only
        communicates every H inner steps, tearing the group down and
        re-initialising on a fresh rendezvous at the next outer sync is cheap
        (one bootstrap per membership change, amortized over H steps). The
        scheduler names the surviving/joining ranks and a fresh port; every
        member calls reform() with its new rank before the next outer sync.
        (The reference's star-topology PS achieves the same via its mutable
        member set — parameter_server.rs round semantics.)
        """
        if dist.is_initialized():
            dist.destroy_process_group()
        self.rank = rank
        self.world_size = world_size
        if world_size > 1:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
            dist.init_process_group(
                backend=backend,
                rank=rank,
                world_size=world_size,
                init_method=f"tcp://{master_addr}:{master_port}",
                timeout=datetime.timedelta(seconds=timeout_s),
            )
            self.backend = backend
        else:
            self.backend = "none"

    def shutdown(self) -> None:
        if dist.is_initialized():
            dist.destroy_process_group()
