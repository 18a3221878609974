"""Microbench for the fp8 cast+transpose kernel (rocprof/PMC target)."""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def main():
    from hypha_amd import _C

    dev = "cuda:0"
    R, C = (int(sys.argv[1]), int(sys.argv[2])) if len(sys.argv) > 2 else (12288, 4096)
    iters = int(sys.argv[3]) if len(sys.argv) > 3 else 50
    x = torch.randn(R, C, device=dev).bfloat16()
    scale = torch.tensor([0.05], dtype=torch.float32, device=dev)
    amax = torch.zeros(_C.fp8_cast_grid_size(R, C), dtype=torch.float32,
                       device=dev)
    for _ in range(5):
        _C.fp8_cast_transpose(x, scale, amax)
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(iters):
        _C.fp8_cast_transpose(x, scale, amax)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t) / iters
    print(f"{R}x{C}: {dt*1e6:.1f} us  {R*C*4/1e9/dt:.0f} GB/s")


if __name__ == "__main__":
    main()
