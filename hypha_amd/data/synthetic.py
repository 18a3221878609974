"""Synthetic token data + SafeTensors slice files.

Benchmarks run on synthetic data (no network for datasets — BASELINE.json);
the data-node path serves SafeTensors slice files exactly like the reference
data node (/root/reference/crates/data/src/bin/hypha-data.rs:150-209: one
file per slice, announced by count, served by index).
"""

from __future__ import annotations

import os

import torch


class SyntheticTokens:
    """Deterministic random token stream; per-rank seed offset so DiLoCo
    workers see disjoint data, like the scheduler's slice assignment."""

    def __init__(self, vocab_size: int, seq_len: int, batch_size: int, seed: int = 1234, rank: int = 0):
        self.vocab_size = vocab_size
        self.seq_len = seq_len
        self.batch_size = batch_size
        self.gen = torch.Generator().manual_seed(seed + 7919 * rank)

    def next_batch(self) -> tuple[torch.Tensor, torch.Tensor]:
        ids = torch.randint(
            0, self.vocab_size, (self.batch_size, self.seq_len), generator=self.gen
        )
        return ids, ids.clone()  # causal LM: labels are the inputs (shifted in-model)


def write_slice_files(
    out_dir: str, dataset: str, num_slices: int, samples_per_slice: int,
    vocab_size: int, seq_len: int, seed: int = 0,
) -> list[str]:
    """Materialise a dataset as SafeTensors slice files (data-node format)."""
    from safetensors.torch import save_file

    os.makedirs(out_dir, exist_ok=True)
    paths = []
    gen = torch.Generator().manual_seed(seed)
    for i in range(num_slices):
        ids = torch.randint(0, vocab_size, (samples_per_slice, seq_len), generator=gen)
        path = os.path.join(out_dir, f"{dataset}-{i:05d}.safetensors")
        save_file({"input_ids": ids}, path)
        paths.append(path)
    return paths


def load_slice(path: str) -> torch.Tensor:
    from safetensors.torch import load_file

    return load_file(path)["input_ids"]
