"""Checkpoint/resume in SafeTensors format.

The reference's de-facto durable state is SafeTensors files
(`0_global_weights.pt`, per-round `{n}_local_gradients.pt`, PS `momentum` /
`avg-final` — training.py:61-63, parameter_server.rs:393-398) with no actual
resume. Here: {theta_global, outer momentum, inner AdamW state, counters} as
safetensors + a JSON manifest, and real resume (SURVEY.md §5 Checkpoint).
"""

from __future__ import annotations

import json
import os

import torch
from safetensors.torch import load_file, save_file

MANIFEST = "manifest.json"


def _fsync_dir_tree(path: str) -> None:
    """fsync every file in `path` and the directory itself so the rename that
    follows publishes fully-durable contents (a power loss after the swap must
    not leave a truncated 'good' checkpoint)."""
    for root, _dirs, files in os.walk(path):
        for name in files:
            fd = os.open(os.path.join(root, name), os.O_RDONLY)
            try:
                os.fsync(fd)
            finally:
                os.close(fd)
        fd = os.open(root, os.O_RDONLY)
        try:
            os.fsync(fd)
        finally:
            os.close(fd)


def _fsync_parent(path: str) -> None:
    parent = os.path.dirname(os.path.abspath(path)) or "."
    fd = os.open(parent, os.O_RDONLY)
    try:
        os.fsync(fd)
    finally:
        os.close(fd)


def save_checkpoint(worker, out_dir: str) -> None:
    """Save a DiLoCoWorker's (or LeanDiLoCoWorker's) full training state.

    Crash-safe: everything is written into `<out_dir>.tmp` first (manifest
    last), then swapped in with the previous checkpoint briefly parked at
    `<out_dir>.bak` — `load_checkpoint` falls back to `.bak` if a crash
    struck mid-swap, so a partial save never destroys the last good state.
    """
    out_dir = out_dir.rstrip("/")
    tmp = out_dir + ".tmp"
    import shutil

    shutil.rmtree(tmp, ignore_errors=True)
    if hasattr(worker, "fp"):
        _save_full(worker, tmp)
    else:  # lean engine
        _save_lean(worker, tmp)
    _fsync_dir_tree(tmp)
    bak = out_dir + ".bak"
    shutil.rmtree(bak, ignore_errors=True)
    if os.path.isdir(out_dir):
        os.rename(out_dir, bak)
    os.rename(tmp, out_dir)
    _fsync_parent(out_dir)  # make both renames durable before dropping .bak
    shutil.rmtree(bak, ignore_errors=True)


def _save_full(worker, out_dir: str) -> None:
    os.makedirs(out_dir, exist_ok=True)
    fp = worker.fp
    save_file({"theta_global": fp.theta0.cpu()}, os.path.join(out_dir, "0_global_weights.safetensors"))
    state = {
        "outer_momentum": fp.outer_momentum.cpu(),
        "master": fp.master.cpu(),
    }
    if getattr(fp, "state_bits", 32) == 8:
        state.update(m8=fp.m8.cpu(), v8=fp.v8.cpu(),
                     m_scale=fp.m_scale.cpu(), v_scale=fp.v_scale.cpu())
    else:
        state.update(exp_avg=fp.exp_avg.cpu(), exp_avg_sq=fp.exp_avg_sq.cpu())
    save_file(state, os.path.join(out_dir, "optimizer_state.safetensors"))
    manifest = {
        "format": "hypha_amd.checkpoint.v1",
        "numel": fp.numel,
        "inner_step_count": worker.inner_step_count,
        "round": worker.round,
        "steps_in_round": worker.steps_in_round,
        "h": worker.cfg.h,
        "outer_lr": worker.cfg.outer.lr,
        "outer_momentum": worker.cfg.outer.momentum,
    }
    with open(os.path.join(out_dir, MANIFEST), "w") as f:
        json.dump(manifest, f, indent=2)


def _save_lean(worker, out_dir: str) -> None:
    os.makedirs(out_dir, exist_ok=True)
    save_file({"theta_global": worker.theta0_host.clone()},
              os.path.join(out_dir, "0_global_weights.safetensors"))
    opt_state = {
        "outer_momentum": worker.outer_m_host.clone(),
        "params": worker.flat.cpu(),
        "m8": worker.m8.cpu(),
        "v8": worker.v8.cpu(),
        "m_scale": worker.m_scale.cpu(),
        "v_scale": worker.v_scale.cpu(),
    }
    if getattr(worker, "fp8_numel", 0):
        opt_state.update({
            "fp8_w8": worker.flat_w8.cpu(),
            "fp8_wscale": worker.flat_wscale.cpu(),
            "fp8_m8": worker.m8_f.cpu(),
            "fp8_v8": worker.v8_f.cpu(),
            "fp8_m_scale": worker.ms_f.cpu(),
            "fp8_v_scale": worker.vs_f.cpu(),
            "fp8_theta0": worker.theta0_fp8_host.clone(),
            "fp8_outer_momentum": worker.outer_m_fp8_host.clone(),
        })
    save_file(opt_state, os.path.join(out_dir, "optimizer_state.safetensors"))
    manifest = {
        "format": "hypha_amd.checkpoint.lean.v1",
        "numel": worker.numel,
        "fp8_numel": getattr(worker, "fp8_numel", 0),
        "inner_step_count": worker.inner_step_count,
        "round": worker.round,
        "steps_in_round": worker.steps_in_round,
        "h": worker.cfg.h,
        "outer_lr": worker.cfg.outer.lr,
        "outer_momentum": worker.cfg.outer.momentum,
    }
    with open(os.path.join(out_dir, MANIFEST), "w") as f:
        json.dump(manifest, f, indent=2)


def load_checkpoint(worker, ckpt_dir: str) -> dict:
    """Restore a DiLoCoWorker's state in place; returns the manifest."""
    ckpt_dir = ckpt_dir.rstrip("/")
    if not os.path.exists(os.path.join(ckpt_dir, MANIFEST)) and os.path.exists(
        os.path.join(ckpt_dir + ".bak", MANIFEST)
    ):
        ckpt_dir = ckpt_dir + ".bak"  # crash mid-swap: last good checkpoint
    with open(os.path.join(ckpt_dir, MANIFEST)) as f:
        manifest = json.load(f)
    if manifest.get("format", "").startswith("hypha_amd.checkpoint.lean"):
        return _load_lean(worker, ckpt_dir, manifest)
    if manifest["numel"] != worker.fp.numel:
        raise ValueError(
            f"checkpoint numel {manifest['numel']} != model numel {worker.fp.numel}"
        )
    fp = worker.fp
    gw = load_file(os.path.join(ckpt_dir, "0_global_weights.safetensors"))
    fp.theta0.copy_(gw["theta_global"].to(fp.theta0.device))
    opt = load_file(os.path.join(ckpt_dir, "optimizer_state.safetensors"))
    fp.outer_momentum.copy_(opt["outer_momentum"].to(fp.master.device))
    if "m8" in opt:
        fp.m8.copy_(opt["m8"].to(fp.master.device))
        fp.v8.copy_(opt["v8"].to(fp.master.device))
        fp.m_scale.copy_(opt["m_scale"].to(fp.master.device))
        fp.v_scale.copy_(opt["v_scale"].to(fp.master.device))
    else:
        fp.exp_avg.copy_(opt["exp_avg"].to(fp.master.device))
        fp.exp_avg_sq.copy_(opt["exp_avg_sq"].to(fp.master.device))
    fp.master.copy_(opt["master"].to(fp.master.device))
    fp.flat.copy_(fp.master)
    worker.inner_step_count = manifest["inner_step_count"]
    worker.round = manifest["round"]
    worker.steps_in_round = manifest["steps_in_round"]
    return manifest


def _load_lean(worker, ckpt_dir: str, manifest: dict) -> dict:
    if manifest["numel"] != worker.numel:
        raise ValueError("checkpoint numel mismatch")
    gw = load_file(os.path.join(ckpt_dir, "0_global_weights.safetensors"))
    worker.theta0_host.copy_(gw["theta_global"])
    opt = load_file(os.path.join(ckpt_dir, "optimizer_state.safetensors"))
    worker.outer_m_host.copy_(opt["outer_momentum"])
    worker.flat.copy_(opt["params"].to(worker.device))
    worker.m8.copy_(opt["m8"].to(worker.device))
    worker.v8.copy_(opt["v8"].to(worker.device))
    worker.m_scale.copy_(opt["m_scale"].to(worker.device))
    worker.v_scale.copy_(opt["v_scale"].to(worker.device))
    if manifest.get("fp8_numel", 0):
        if manifest["fp8_numel"] != worker.fp8_numel:
            raise ValueError("checkpoint fp8 numel mismatch")
        worker.flat_w8.copy_(opt["fp8_w8"].to(worker.device))
        worker.flat_wscale.copy_(opt["fp8_wscale"].to(worker.device))
        worker.m8_f.copy_(opt["fp8_m8"].to(worker.device))
        worker.v8_f.copy_(opt["fp8_v8"].to(worker.device))
        worker.ms_f.copy_(opt["fp8_m_scale"].to(worker.device))
        worker.vs_f.copy_(opt["fp8_v_scale"].to(worker.device))
        worker.theta0_fp8_host.copy_(opt["fp8_theta0"])
        worker.outer_m_fp8_host.copy_(opt["fp8_outer_momentum"])
    worker.inner_step_count = manifest["inner_step_count"]
    worker.round = manifest["round"]
    worker.steps_in_round = manifest["steps_in_round"]
    return manifest
