"""Loss-function and LR-schedule wire registries.

Parity with the reference executor's hyperparameter surface:
  * losses — executors/accelerate/.../utils.py:76-87 and the `Loss` wire enum
    (crates/messages/src/lib.rs:662-670): l1, mse, cross-entropy,
    bce-with-logits, kl-div (kebab-case wire names).
  * LR schedules — utils.py:90-106 and `Scheduler` (lib.rs:672-696):
    cosine-with-warmup / linear-with-warmup / wsd, absent -> constant.
    Here a wire schedule dict maps onto `InnerOptConfig` fields consumed by
    `hypha_amd.parallel.lr_at` (the fused-optimizer path takes the LR as a
    scalar per step, so schedules stay host-side like the reference's K3).
"""

from __future__ import annotations

from typing import Callable

import torch

_LOSSES: dict[str, Callable[[], torch.nn.Module]] = {
    "l1": torch.nn.L1Loss,
    "mse": torch.nn.MSELoss,
    "cross-entropy": torch.nn.CrossEntropyLoss,
    "bce-with-logits": torch.nn.BCEWithLogitsLoss,
    "kl-div": torch.nn.KLDivLoss,
}


def get_loss_fn(name: str) -> torch.nn.Module:
    try:
        return _LOSSES[name]()
    except KeyError:
        raise ValueError(
            f"loss {name!r} not supported; available: {sorted(_LOSSES)}"
        ) from None


def apply_wire_schedule(inner_cfg, schedule: dict | None) -> None:
    """Apply a wire `Scheduler` dict onto an InnerOptConfig in place.

    {"type": "cosine-with-warmup", "warmup_steps": W, "training_steps": T}
    {"type": "linear-with-warmup", "warmup_steps": W, "training_steps": T}
    {"type": "wsd", "warmup_steps": W, "decay_step": D}
    None / missing type -> constant (utils.py:91-92).
    """
    if not schedule or not schedule.get("type"):
        inner_cfg.schedule = "constant"
        return
    kind = schedule["type"]
    inner_cfg.warmup_steps = int(schedule.get("warmup_steps", inner_cfg.warmup_steps))
    if kind == "cosine-with-warmup":
        inner_cfg.schedule = "cosine"
        inner_cfg.total_steps = int(schedule["training_steps"])
    elif kind == "linear-with-warmup":
        inner_cfg.schedule = "linear"
        inner_cfg.total_steps = int(schedule["training_steps"])
    elif kind == "wsd":
        inner_cfg.schedule = "wsd"
        # reference wsd takes the step where decay BEGINS; our wsd decays over
        # the last 10% of (total_steps - warmup) — solve total so decay starts
        # exactly at decay_step: warmup + 0.9*(total-warmup) = decay_step
        decay_step = int(schedule["decay_step"])
        w = inner_cfg.warmup_steps
        inner_cfg.total_steps = max(decay_step + 1, w + round((decay_step - w) / 0.9))
    else:
        raise ValueError(f"LR scheduler {kind!r} not supported")
