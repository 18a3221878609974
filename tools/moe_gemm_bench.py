"""Grouped-GEMM kernel vs per-expert hipBLASLt at MoE shapes."""
import sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch

def bench(fn, iters=20, warm=5):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

def main():
    from hypha_amd import _C
    E, K, N = 8, 4096, 14336
    for T in (512, 4096, 16384):
        x = torch.randn(T, K, device="cuda").bfloat16()
        w = torch.randn(E, N, K, device="cuda").bfloat16()
        cnt = T // E
        off = torch.arange(0, T + 1, cnt, dtype=torch.int32)
        t_native = bench(lambda: _C.grouped_gemm(x, w, off))
        def torch_loop():
            outs = [x[g*cnt:(g+1)*cnt] @ w[g].t() for g in range(E)]
            return torch.cat(outs)
        t_torch = bench(torch_loop)
        fl = 2 * T * K * N
        print(f"T={T:6d}: native {t_native*1e3:7.3f} ms ({fl/t_native/1e12:6.1f} TF) | "
              f"hipBLASLt loop {t_torch*1e3:7.3f} ms ({fl/t_torch/1e12:6.1f} TF)")

main()
