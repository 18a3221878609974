"""Probe fp8 (OCP e4m3) GEMM support on this torch/ROCm build."""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch, time

def main():
    print("float8_e4m3fn:", hasattr(torch, "float8_e4m3fn"))
    a = torch.randn(512, 256, device="cuda").bfloat16()
    b = torch.randn(1024, 256, device="cuda").bfloat16()
    try:
        a8 = a.to(torch.float8_e4m3fn)
        b8 = b.to(torch.float8_e4m3fn)
        sa = torch.tensor(1.0, device="cuda")
        out = torch._scaled_mm(a8, b8.t(), scale_a=sa, scale_b=sa, out_dtype=torch.bfloat16)
        ref = a.float() @ b.float().t()
        rel = (out.float() - ref).norm() / ref.norm()
        print("scaled_mm ok, rel err:", float(rel))
        # timing at a bench-relevant shape
        M, K, N = 8192, 4096, 4096
        x = torch.randn(M, K, device="cuda").bfloat16()
        w = torch.randn(N, K, device="cuda").bfloat16()
        x8, w8 = x.to(torch.float8_e4m3fn), w.to(torch.float8_e4m3fn)
        for args, tag in ((("bf16",), "bf16"), (("fp8",), "fp8")):
            def f():
                if tag == "bf16":
                    return x @ w.t()
                return torch._scaled_mm(x8, w8.t(), scale_a=sa, scale_b=sa, out_dtype=torch.bfloat16)
            for _ in range(5): f()
            torch.cuda.synchronize(); t0 = time.perf_counter()
            for _ in range(20): f()
            torch.cuda.synchronize()
            dt = (time.perf_counter()-t0)/20
            print(f"{tag}: {dt*1e3:.3f} ms  {2*M*K*N/dt/1e12:.0f} TF/s")
    except Exception as e:
        print("scaled_mm FAILED:", repr(e))

main()
