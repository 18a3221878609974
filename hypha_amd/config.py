"""Layered configuration: defaults -> config file (TOML/JSON) -> HYPHA_* env
-> explicit overrides, with validation and commented example generation.

Parity with the reference's figment-based config crate
(/root/reference/crates/config/src/lib.rs: TOML -> HYPHA_* env -> CLI
layering :565-613, validate() diagnostics :403-451, `init` example
generation via doc comments :544).
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field, fields
from typing import Any


class ConfigError(ValueError):
    """Validation error with the offending key (miette-style provenance)."""

    def __init__(self, key: str, message: str, source: str = ""):
        self.key = key
        self.source = source
        super().__init__(f"config key '{key}': {message}" + (f" (from {source})" if source else ""))


@dataclass
class JobConfig:
    """DiLoCo job specification (scheduler_config.rs:52-105 analogue)."""

    model: str = "llama3-8b"  # model registry name
    dataset: str = "synthetic"  # dataset name announced by the data node
    num_workers: int = 8  # DiLoCo worker peers (GPUs)
    update_rounds: int = 100  # outer rounds
    avg_samples_between_updates: int = 1200  # samples per round per worker
    batch_size: int = 4  # per-worker batch (sequences)
    seq_len: int = 2048
    inner_lr: float = 4e-4  # inner AdamW learning rate
    inner_beta1: float = 0.9
    inner_beta2: float = 0.95
    inner_weight_decay: float = 0.1
    outer_lr: float = 0.7  # outer Nesterov (reference default)
    outer_momentum: float = 0.9
    lr_schedule: str = "cosine"  # constant | cosine | linear | wsd
    grad_clip: float = 1.0
    worker_bid: float = 1.0  # auction bid price
    worker_max_price: float = 10.0
    checkpoint_dir: str = ""
    checkpoint_every_rounds: int = 0  # 0 = disabled
    max_batch_size: int = 600  # cap for capacity-proportional batches
    weighted_aggregation: bool = False  # weight deltas by sample counts
    job_type: str = "diloco"  # diloco | generate (inference dispatch)
    max_new_tokens: int = 16  # generate mode
    num_batches: int = 2  # generate mode: batches per worker

    def validate(self) -> "JobConfig":
        if self.num_workers < 1:
            raise ConfigError("num_workers", "must be >= 1")
        if self.update_rounds < 1:
            raise ConfigError("update_rounds", "must be >= 1")
        if self.lr_schedule not in ("constant", "cosine", "linear", "wsd"):
            raise ConfigError("lr_schedule", f"unknown schedule {self.lr_schedule!r}")
        if not (0 <= self.outer_momentum < 1):
            raise ConfigError("outer_momentum", "must be in [0, 1)")
        if self.batch_size < 1 or self.seq_len < 1:
            raise ConfigError("batch_size/seq_len", "must be positive")
        if self.job_type not in ("diloco", "generate"):
            raise ConfigError("job_type", f"unknown job type {self.job_type!r}")
        return self


def _coerce(value: str, target_type) -> Any:
    if target_type is bool:
        return value.lower() in ("1", "true", "yes")
    return target_type(value)


def load_job_config(path: str | None = None, env: dict | None = None,
                    **overrides) -> JobConfig:
    """Layer: defaults -> file -> HYPHA_* env -> overrides, then validate."""
    env = os.environ if env is None else env
    data: dict[str, Any] = {}
    source: dict[str, str] = {}
    if path:
        if path.endswith(".toml"):
            try:
                import tomli as toml_mod
            except ImportError:  # py>=3.11
                import tomllib as toml_mod
            with open(path, "rb") as f:
                data.update(toml_mod.load(f))
        else:
            with open(path) as f:
                data.update(json.load(f))
        for k in data:
            source[k] = path
    for f_ in fields(JobConfig):
        env_key = f"HYPHA_{f_.name.upper()}"
        if env_key in env:
            data[f_.name] = _coerce(env[env_key], f_.type if isinstance(f_.type, type) else type(getattr(JobConfig, f_.name, "")))
            source[f_.name] = f"env:{env_key}"
    data.update({k: v for k, v in overrides.items() if v is not None})
    known = {f_.name for f_ in fields(JobConfig)}
    unknown = set(data) - known
    if unknown:
        k = sorted(unknown)[0]
        raise ConfigError(k, "unknown key", source.get(k, ""))
    try:
        return JobConfig(**data).validate()
    except TypeError as e:
        raise ConfigError("<root>", str(e))


def example_config() -> str:
    """Commented example (the reference's `init` subcommand output)."""
    lines = ["# hypha_amd DiLoCo job configuration (JSON; TOML also accepted)", "{"]
    doc = {
        "model": "model registry name (llama3-8b, llama3-70b, mixtral-8x7b, gpt2-small)",
        "dataset": "dataset name announced by the data node",
        "num_workers": "DiLoCo worker peers (one per GPU)",
        "update_rounds": "outer synchronization rounds",
        "avg_samples_between_updates": "samples per round per worker (H*batch)",
        "batch_size": "per-worker batch in sequences",
        "seq_len": "sequence length",
        "inner_lr": "inner AdamW learning rate",
        "outer_lr": "outer Nesterov learning rate",
        "outer_momentum": "outer Nesterov momentum",
        "lr_schedule": "constant | cosine | linear | wsd",
    }
    cfg = JobConfig()
    items = []
    for f_ in fields(JobConfig):
        v = json.dumps(getattr(cfg, f_.name))
        comment = doc.get(f_.name, "")
        items.append(f'  "{f_.name}": {v}' + (f"  // {comment}" if comment else ""))
    lines.extend(",\n".join(items).split("\n"))
    lines.append("}")
    return "\n".join(lines)
