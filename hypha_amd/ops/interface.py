"""Public op API: autograd wrappers that dispatch HIP kernels (GPU) or the
PyTorch reference path (CPU).

The HIP extension ABI (hypha_amd._C):
  rmsnorm_fwd(x2d, w, eps) -> (y, rstd)
  rmsnorm_bwd(dy, x2d, w, rstd) -> (dx, dw)
  rope_fwd_ex(q, k, cos, sin, inverse, layout) -> (q_out, k_out)
  swiglu_fwd(gate, up) -> out ; swiglu_bwd(dout, gate, up) -> (dgate, dup)
  ce_fwd(logits2d, targets) -> (loss_sum, lse, n_valid)    # fp32 scalars/rows
  ce_bwd_(logits2d, targets, lse, scale) -> dlogits        # overwrites logits
  attn_fwd_ex(q, k, v, causal, layout) -> (o, lse)         # layout bhsd|bshd
  attn_bwd_ex(q, k, v, o, do, lse, causal, layout) -> (dq, dk, dv)
  adamw_step_(master, param, grad, m, v, lr, b1, b2, eps, wd, step)
  adamw8_step_(master, param, grad, m8, v8, m_scale, v_scale, ...)  # 8-bit state
  nesterov_step_(master, delta_bf16, momentum, lr, mu)
  extract_delta(master, theta0, out_bf16)
  grad_norm_sq(flat_bf16) -> fp32 scalar
"""

from __future__ import annotations

import torch

from . import reference
from . import use_native


def _c():
    from hypha_amd import _C

    return _C


# ---------------------------------------------------------------------------
# RMSNorm
# ---------------------------------------------------------------------------


class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        shape = x.shape
        x2d = x.reshape(-1, shape[-1]).contiguous()
        y, rstd = _c().rmsnorm_fwd(x2d, weight, eps)
        ctx.save_for_backward(x2d, weight, rstd)
        ctx.shape = shape
        return y.view(shape)

    @staticmethod
    def backward(ctx, dy):
        x2d, weight, rstd = ctx.saved_tensors
        dx, dw = _c().rmsnorm_bwd(dy.reshape(x2d.shape).contiguous(), x2d, weight, rstd)
        return dx.view(ctx.shape), dw.to(weight.dtype), None


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if use_native(x):
        return _RMSNorm.apply(x, weight, eps)
    return reference.rmsnorm(x, weight, eps)


class _AddRMSNorm(torch.autograd.Function):
    """Fused h = a + b; y = rmsnorm(h): the residual add is computed inside
    the norm kernel (one fewer elementwise pass per transformer sub-block);
    backward fuses the residual gradient into the dx pass."""

    @staticmethod
    def forward(ctx, a2d, b2d, weight, eps):
        h, y, rstd = _c().add_rmsnorm_fwd(a2d, b2d, weight, eps)
        ctx.save_for_backward(h, weight, rstd)
        return h, y

    @staticmethod
    def backward(ctx, dh, dy):
        h, weight, rstd = ctx.saved_tensors
        if dh is None:
            dx, dw = _c().rmsnorm_bwd(dy.contiguous(), h, weight, rstd)
        else:
            dx, dw = _c().add_rmsnorm_bwd(dy.contiguous(), dh.contiguous(), h,
                                          weight, rstd)
        return dx, dx, dw.to(weight.dtype), None


def add_rmsnorm(a: torch.Tensor, b: torch.Tensor, weight: torch.Tensor,
                eps: float = 1e-5):
    """Returns (h, y) with h = a + b and y = rmsnorm(h), fused on GPU."""
    if use_native(a):
        shape = a.shape
        h, y = _AddRMSNorm.apply(a.reshape(-1, shape[-1]).contiguous(),
                                 b.reshape(-1, shape[-1]).contiguous(),
                                 weight, eps)
        return h.view(shape), y.view(shape)
    h = a + b
    return h, reference.rmsnorm(h, weight, eps)


# ---------------------------------------------------------------------------
# RoPE on q and k together (one fused kernel launch)
# ---------------------------------------------------------------------------


class _RoPE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, cos, sin, layout):
        qo, ko = _c().rope_fwd_ex(q.contiguous(), k.contiguous(), cos, sin, False, layout)
        ctx.save_for_backward(cos, sin)
        ctx.layout = layout
        return qo, ko

    @staticmethod
    def backward(ctx, dq, dk):
        cos, sin = ctx.saved_tensors
        dqo, dko = _c().rope_fwd_ex(dq.contiguous(), dk.contiguous(), cos, sin, True,
                                    ctx.layout)
        return dqo, dko, None, None, None


def apply_rope_qk(q, k, cos, sin, layout: str = "bhsd"):
    """layout 'bhsd': q [B,Hq,S,D]; 'bshd': q [B,S,Hq,D] (the projection's
    natural layout — avoids transpose copies). cos/sin [S_max, D/2] fp32."""
    if use_native(q):
        return _RoPE.apply(q, k, cos, sin, layout)
    if layout == "bshd":
        qo = reference.apply_rope(q.transpose(1, 2), cos, sin).transpose(1, 2)
        ko = reference.apply_rope(k.transpose(1, 2), cos, sin).transpose(1, 2)
        return qo, ko
    return reference.apply_rope(q, cos, sin), reference.apply_rope(k, cos, sin)


# ---------------------------------------------------------------------------
# LayerNorm + tanh-GELU (the GPT-2 block's norm/activation)
# ---------------------------------------------------------------------------


class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        shape = x.shape
        x2d = x.reshape(-1, shape[-1]).contiguous()
        y, mean, rstd = _c().layernorm_fwd(x2d, weight, bias, eps)
        ctx.save_for_backward(x2d, weight, mean, rstd)
        ctx.shape = shape
        return y.view(shape)

    @staticmethod
    def backward(ctx, dy):
        x2d, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = _c().layernorm_bwd(dy.reshape(x2d.shape).contiguous(), x2d,
                                        weight, mean, rstd)
        return dx.view(ctx.shape), dw.to(weight.dtype), db.to(weight.dtype), None


def layernorm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
              eps: float = 1e-5) -> torch.Tensor:
    if use_native(x):
        return _LayerNorm.apply(x, weight, bias, eps)
    return torch.nn.functional.layer_norm(x, (x.shape[-1],), weight, bias, eps)


class _Gelu(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = x.contiguous()
        ctx.save_for_backward(x)
        return _c().gelu_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        return _c().gelu_bwd(dy.contiguous(), x)


def gelu(x: torch.Tensor) -> torch.Tensor:
    """tanh-approximation GELU (GPT-2's activation)."""
    if use_native(x):
        return _Gelu.apply(x)
    return torch.nn.functional.gelu(x, approximate="tanh")


# ---------------------------------------------------------------------------
# SwiGLU
# ---------------------------------------------------------------------------


class _SwiGLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate, up):
        gate = gate.contiguous()
        up = up.contiguous()
        ctx.save_for_backward(gate, up)
        return _c().swiglu_fwd(gate, up)

    @staticmethod
    def backward(ctx, dout):
        gate, up = ctx.saved_tensors
        dgate, dup = _c().swiglu_bwd(dout.contiguous(), gate, up)
        return dgate, dup


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    if use_native(gate):
        return _SwiGLU.apply(gate, up)
    return reference.swiglu(gate, up)


# ---------------------------------------------------------------------------
# Flash attention (causal, GQA)
# ---------------------------------------------------------------------------


class _FlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal, layout):
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        o, lse = _c().attn_fwd_ex(q, k, v, causal, layout)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        ctx.layout = layout
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = _c().attn_bwd_ex(q, k, v, o, do.contiguous(), lse, ctx.causal,
                                      ctx.layout)
        return dq, dk, dv, None, None


def flash_attention(q, k, v, causal: bool = True, layout: str = "bhsd") -> torch.Tensor:
    """Causal GQA attention. layout 'bhsd': q [B,Hq,S,D]; 'bshd': q [B,S,Hq,D]
    (no transpose copies around the projections)."""
    if use_native(q):
        return _FlashAttention.apply(q, k, v, causal, layout)
    if layout == "bshd":
        o = reference.attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2), causal
        )
        return o.transpose(1, 2)
    return reference.attention(q, k, v, causal)


# ---------------------------------------------------------------------------
# KV-cache decode attention (flash-decoding; serving path)
# ---------------------------------------------------------------------------


def attn_decode(q, kcache, vcache, t: int, kscale=None, vscale=None) -> torch.Tensor:
    """Single-token attention over an appended KV cache.

    q [B, Hq, D] bf16; kcache/vcache [B, T_alloc, Hkv, D] bf16 OR fp8-e4m3
    with per-row scales (bshd); t = valid length including the current
    token. Returns [B, Hq, D] bf16. Native split-KV kernel
    (ops/hip/decode.hip); fp32 reference otherwise.
    """
    quant = kcache.dtype == torch.float8_e4m3fn
    if use_native(q) and q.shape[-1] in (64, 128) and q.shape[1] <= 8 * kcache.shape[2]:
        if quant:
            return _c().attn_decode_fp8(q, kcache, vcache, kscale, vscale, t)
        return _c().attn_decode(q, kcache, vcache, t)
    # reference path (CPU, or head dims the kernel doesn't cover):
    # plain masked softmax over the valid prefix
    if quant:
        kc = kcache[:, :t].float() * kscale[:, :t].unsqueeze(-1).float()
        vc = vcache[:, :t].float() * vscale[:, :t].unsqueeze(-1).float()
    else:
        kc = kcache[:, :t].float()
        vc = vcache[:, :t].float()
    rep = q.shape[1] // kcache.shape[2]
    kh = kc.permute(0, 2, 1, 3).repeat_interleave(rep, dim=1)
    vh = vc.permute(0, 2, 1, 3).repeat_interleave(rep, dim=1)
    scores = torch.einsum("bhd,bhtd->bht", q.float(), kh) / (q.shape[-1] ** 0.5)
    p = torch.softmax(scores, dim=-1)
    o = torch.einsum("bht,bhtd->bhd", p, vh)
    return o.to(q.dtype)


def linear_skinny(x2d: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    """Inference linear for decode shapes: out = x2d @ weight.T with the
    weight-bandwidth-bound skinny-M kernel when M <= 8 (ops/hip/
    skinny_gemm.hip); hipBLASLt otherwise. No autograd."""
    if (use_native(x2d) and x2d.shape[0] <= 8
            and weight.dtype == torch.bfloat16):
        return _c().skinny_gemm(x2d.contiguous(), weight)
    return x2d @ weight.t()


# ---------------------------------------------------------------------------
# Cross-entropy over vocab (memory-frugal: backward overwrites the logits)
# ---------------------------------------------------------------------------


class _CrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits2d, targets):
        loss_sum, lse, n_valid = _c().ce_fwd(logits2d, targets)
        ctx.save_for_backward(logits2d, targets, lse)
        ctx.n_valid = max(int(n_valid), 1)
        return loss_sum / ctx.n_valid

    @staticmethod
    def backward(ctx, dloss):
        logits2d, targets, lse = ctx.saved_tensors
        scale = float(dloss) / ctx.n_valid
        dlogits = _c().ce_bwd_(logits2d, targets, lse, scale)
        return dlogits, None


def cross_entropy_loss(logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
    """logits [..., V], targets [...] int64 with -100 = ignore. Mean loss."""
    v = logits.shape[-1]
    logits2d = logits.reshape(-1, v)
    t = targets.reshape(-1)
    if use_native(logits):
        return _CrossEntropy.apply(logits2d.contiguous(), t.contiguous())
    return reference.cross_entropy(logits2d, t)


# ---------------------------------------------------------------------------
# Fused optimizers (no autograd; flat-buffer in-place)
# ---------------------------------------------------------------------------


@torch.no_grad()
def fused_adamw(master, param, grad, exp_avg, exp_avg_sq, *, lr, beta1, beta2, eps, weight_decay, step):
    if use_native(master):
        _c().adamw_step_(master, param, grad, exp_avg, exp_avg_sq, lr, beta1, beta2, eps, weight_decay, step)
    else:
        reference.adamw_step(
            master, param, grad, exp_avg, exp_avg_sq,
            lr=lr, beta1=beta1, beta2=beta2, eps=eps, weight_decay=weight_decay, step=step,
        )


@torch.no_grad()
def fused_nesterov(master, delta, momentum, *, lr, mu):
    if use_native(master):
        _c().nesterov_step_(master, delta, momentum, lr, mu)
    else:
        reference.nesterov_outer_step(master, delta, momentum, lr=lr, mu=mu)


@torch.no_grad()
def extract_delta(master: torch.Tensor, theta0: torch.Tensor, out: torch.Tensor) -> torch.Tensor:
    """out (bf16 or fp32) = master - theta0, fused sub+cast on GPU."""
    if use_native(master):
        _c().extract_delta(master, theta0, out)
        return out
    out.copy_(master - theta0)
    return out
