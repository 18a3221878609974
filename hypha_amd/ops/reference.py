"""Pure-PyTorch reference implementations of every hot op.

These are (a) the CPU execution path (no GPU in CI), and (b) the fp32 golden
oracle each HIP kernel's numerics test compares against — the pattern the
reference repo uses for its Rust Nesterov pipeline
(/root/reference/crates/worker/src/executor/parameter_server.rs:448-525,
golden-value test vs torch.optim.SGD(nesterov=True)).
"""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F


# ---------------------------------------------------------------------------
# Normalization
# ---------------------------------------------------------------------------

def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    """RMSNorm: x * rsqrt(mean(x^2) + eps) * weight, computed in fp32."""
    dtype = x.dtype
    x32 = x.float()
    var = x32.pow(2).mean(-1, keepdim=True)
    return (x32 * torch.rsqrt(var + eps)).to(dtype) * weight


# ---------------------------------------------------------------------------
# Rotary position embedding (RoPE) — Llama-3 convention: rotate half pairs
# interleaved as (x[..., :d/2], x[..., d/2:]).
# ---------------------------------------------------------------------------

def rope_cos_sin(
    seq_len: int,
    head_dim: int,
    base: float = 500000.0,
    device=None,
    dtype=torch.float32,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Precomputed cos/sin tables of shape [seq_len, head_dim // 2]."""
    inv_freq = 1.0 / (
        base ** (torch.arange(0, head_dim, 2, device=device, dtype=torch.float32) / head_dim)
    )
    t = torch.arange(seq_len, device=device, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)  # [S, D/2]
    return freqs.cos().to(dtype), freqs.sin().to(dtype)


def apply_rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    """Apply RoPE to x of shape [B, H, S, D]; cos/sin are [S, D/2]."""
    d = x.shape[-1] // 2
    x1, x2 = x[..., :d], x[..., d:]
    c = cos[: x.shape[-2]].view(1, 1, -1, d)
    s = sin[: x.shape[-2]].view(1, 1, -1, d)
    out1 = x1.float() * c - x2.float() * s
    out2 = x2.float() * c + x1.float() * s
    return torch.cat([out1, out2], dim=-1).to(x.dtype)


# ---------------------------------------------------------------------------
# Attention (causal, GQA) — reference is plain SDPA-equivalent math
# ---------------------------------------------------------------------------

def attention(
    q: torch.Tensor,  # [B, Hq, S, D]
    k: torch.Tensor,  # [B, Hkv, S, D]
    v: torch.Tensor,  # [B, Hkv, S, D]
    causal: bool = True,
) -> torch.Tensor:
    b, hq, s, d = q.shape
    hkv = k.shape[1]
    if hkv != hq:
        rep = hq // hkv
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    scale = 1.0 / math.sqrt(d)
    scores = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    if causal:
        mask = torch.triu(
            torch.ones(s, s, dtype=torch.bool, device=q.device), diagonal=1
        )
        scores = scores.masked_fill(mask, float("-inf"))
    p = torch.softmax(scores, dim=-1)
    return torch.matmul(p, v.float()).to(q.dtype)


# ---------------------------------------------------------------------------
# SwiGLU MLP activation
# ---------------------------------------------------------------------------

def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """silu(gate) * up in fp32."""
    return (F.silu(gate.float()) * up.float()).to(gate.dtype)


# ---------------------------------------------------------------------------
# Cross-entropy over vocab (fp32 logits path)
# ---------------------------------------------------------------------------

def cross_entropy(logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
    """Mean CE over all positions; logits [N, V], targets [N] (-100 ignored)."""
    return F.cross_entropy(logits.float(), targets, ignore_index=-100)


# ---------------------------------------------------------------------------
# Fused inner AdamW (decoupled weight decay) over flat fp32 state
# — oracle for the HIP kernel (K2 in SURVEY.md §2.10)
# ---------------------------------------------------------------------------

@torch.no_grad()
def adamw_step(
    master: torch.Tensor,  # fp32 flat master weights, updated in place
    param_bf16: torch.Tensor,  # bf16 working copy, updated in place
    grad: torch.Tensor,  # grad (any float dtype), same numel
    exp_avg: torch.Tensor,  # fp32 m, in place
    exp_avg_sq: torch.Tensor,  # fp32 v, in place
    *,
    lr: float,
    beta1: float = 0.9,
    beta2: float = 0.95,
    eps: float = 1e-8,
    weight_decay: float = 0.1,
    step: int,
) -> None:
    g = grad.float()
    exp_avg.mul_(beta1).add_(g, alpha=1.0 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1.0 - beta2)
    bc1 = 1.0 - beta1 ** step
    bc2 = 1.0 - beta2 ** step
    denom = (exp_avg_sq / bc2).sqrt_().add_(eps)
    master.mul_(1.0 - lr * weight_decay)
    master.addcdiv_(exp_avg, denom, value=-lr / bc1)
    param_bf16.copy_(master)


# ---------------------------------------------------------------------------
# Outer Nesterov (DiLoCo aggregate step) over flat fp32 state
# — oracle for the HIP kernel (K7/K8). Matches the reference parameter
#   server's semantics (parameter_server.rs:386-446): with pseudo-gradient
#   delta = theta_t - theta_0 (the *negative* of a gradient), the update is
#     m      <- mu * m + delta
#     theta  <- theta + lr * (mu * m + delta)
#   which equals torch.optim.SGD(lr, momentum=mu, nesterov=True) applied to
#   gradient g = -delta.
# ---------------------------------------------------------------------------

@torch.no_grad()
def nesterov_outer_step(
    master: torch.Tensor,  # fp32 flat global weights (theta_0), in place
    delta: torch.Tensor,  # averaged pseudo-gradient (theta_t - theta_0)
    momentum: torch.Tensor,  # fp32 outer momentum, in place
    *,
    lr: float,
    mu: float = 0.9,
) -> None:
    d = delta.float()
    momentum.mul_(mu).add_(d)
    master.add_(momentum, alpha=lr * mu).add_(d, alpha=lr)


# ---------------------------------------------------------------------------
# Pseudo-gradient extraction / merge (K4/K5):
#   delta = theta_t - theta_0 ; merge: theta <- theta_prev + delta
# (reference: executors/accelerate/.../utils.py:105-123)
# ---------------------------------------------------------------------------

@torch.no_grad()
def extract_delta(theta_t: torch.Tensor, theta_0: torch.Tensor, out: torch.Tensor) -> torch.Tensor:
    torch.sub(theta_t, theta_0, out=out)
    return out
