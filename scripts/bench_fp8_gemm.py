#!/usr/bin/env python3
"""fp8 vs bf16 GEMM microbench on the FLUX hot shapes (round-2 data)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

def timeit(fn, iters=15, warmup=5):
    for _ in range(warmup): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters

FP8 = torch.float8_e4m3fn
shapes = [(36864, 3072, 9216), (36864, 3072, 12288), (36864, 12288, 3072),
          (32768, 3072, 9216), (32768, 3072, 12288), (36864, 3072, 3072)]
one = torch.ones((), device="cuda")
for (m, k, n) in shapes:
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    a8 = a.to(FP8); w8 = w.to(FP8)
    tf = 2 * m * k * n / 1e12
    t_bf = timeit(lambda: torch.nn.functional.linear(a, w))
    t_f8 = timeit(lambda: torch._scaled_mm(a8, w8.t(), scale_a=one, scale_b=one,
                                           out_dtype=torch.bfloat16))
    # dynamic per-tensor act quant overhead (what the current fp8 mode pays)
    def quant_path():
        s = (a.abs().amax().float() / 448.0).clamp(min=1e-12)
        aq = (a.float() / s).clamp(-448, 448).to(FP8)
        return torch._scaled_mm(aq, w8.t(), scale_a=s, scale_b=one,
                                out_dtype=torch.bfloat16)
    t_q = timeit(quant_path)
    print(f"{m}x{k}x{n}: bf16 {t_bf*1e3:7.3f} ms ({tf/t_bf:6.0f} TF) | "
          f"fp8 {t_f8*1e3:7.3f} ms ({tf/t_f8:6.0f} TF) | "
          f"fp8+dynquant {t_q*1e3:7.3f} ms ({tf/t_q:6.0f} TF)")
