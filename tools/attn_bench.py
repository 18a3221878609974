"""Micro-benchmark for the attention kernels at Llama-3-8B shapes.

Usage: python tools/attn_bench.py [--iters N] [--mode fwd|bwd|both]
Prints TF/s for forward and backward at B4 Hq32 Hkv8 S2048 D128 (causal).
"""

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--mode", default="both")
    p.add_argument("--batch", type=int, default=4)
    p.add_argument("--seq", type=int, default=2048)
    args = p.parse_args()

    from hypha_amd import _C

    B, Hq, Hkv, S, D = args.batch, 32, 8, args.seq, 128
    torch.manual_seed(0)
    dev = "cuda:0"
    q = torch.randn(B, Hq, S, D, device=dev).bfloat16()
    k = torch.randn(B, Hkv, S, D, device=dev).bfloat16()
    v = torch.randn(B, Hkv, S, D, device=dev).bfloat16()
    do = torch.randn(B, Hq, S, D, device=dev).bfloat16()

    o, lse = _C.attn_fwd(q, k, v, True)

    # causal flops: QK^T + PV = 2 * 2 * B*Hq*S^2*D / 2 (triangle)
    fwd_flops = 2 * 2 * B * Hq * S * S * D / 2
    bwd_flops = fwd_flops * 2.5  # 5 matmuls vs 2

    def bench(fn, iters):
        for _ in range(args.warmup):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters

    if args.mode in ("fwd", "both"):
        t = bench(lambda: _C.attn_fwd(q, k, v, True), args.iters)
        print(f"fwd: {t*1e3:.3f} ms  {fwd_flops/t/1e12:.1f} TF/s")
    if args.mode in ("bwd", "both"):
        t = bench(lambda: _C.attn_bwd(q, k, v, o, do, lse, True), args.iters)
        print(f"bwd: {t*1e3:.3f} ms  {bwd_flops/t/1e12:.1f} TF/s")


if __name__ == "__main__":
    main()
