#!/usr/bin/env python3
"""Attention / op microbench on MI355X (run under gpurun).

Reports TF/s for the fused attention kernel at FLUX/SDXL/WAN shapes and
times the per-layer fused ops; also a hipBLASLt bf16 GEMM reference to show
the roofline context. Random data (guide §5.4 rule 25)."""
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from comfyui_parallelanything_amd import ops  # noqa: E402


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def attn_flops(B, H, S, D):
    return 4.0 * B * H * S * S * D  # QK^T + PV, 2 FLOP per MAC


def main():
    assert torch.cuda.is_available()
    dev = "cuda"
    print("== attention (bf16, random data) ==")
    for name, (B, H, S, D) in {
        "flux_b8": (8, 24, 4608, 128),
        "flux_b1": (1, 24, 4608, 128),
        "sdxl_mid": (4, 20, 1024, 64),
        "square_2k": (16, 16, 2048, 128),
    }.items():
        q = torch.randn(B, H, S, D, device=dev, dtype=torch.bfloat16)
        k = torch.randn_like(q)
        v = torch.randn_like(q)
        dt = timeit(lambda: ops.attention(q, k, v), iters=10)
        tf = attn_flops(B, H, S, D) / dt / 1e12
        print(f"  {name:10s} B{B} H{H} S{S} D{D}: {dt*1e3:8.2f} ms  {tf:7.1f} TF/s")

    print("== hipBLASLt bf16 GEMM reference ==")
    for n in (4096, 8192):
        a = torch.randn(n, n, device=dev, dtype=torch.bfloat16)
        b = torch.randn(n, n, device=dev, dtype=torch.bfloat16)
        dt = timeit(lambda: a @ b, iters=10)
        print(f"  {n}^3: {dt*1e3:8.2f} ms  {2*n**3/dt/1e12:7.1f} TF/s")
    # FLUX linear shapes (M = batch*seq)
    for (m, k_, n) in [(36864, 3072, 3072), (36864, 3072, 12288),
                       (36864, 15360, 3072), (36864, 3072, 21504)]:
        a = torch.randn(m, k_, device=dev, dtype=torch.bfloat16)
        w = torch.randn(n, k_, device=dev, dtype=torch.bfloat16)
        dt = timeit(lambda: torch.nn.functional.linear(a, w), iters=10)
        print(f"  linear {m}x{k_}x{n}: {dt*1e3:8.2f} ms  {2*m*k_*n/dt/1e12:7.1f} TF/s")

    print("== fused elementwise ops (FLUX shapes, batch 8) ==")
    B, S, Dm = 8, 4608, 3072
    x = torch.randn(B, S, Dm, device=dev, dtype=torch.bfloat16)
    sc = torch.randn(B, Dm, device=dev, dtype=torch.bfloat16)
    sh = torch.randn_like(sc)
    dt = timeit(lambda: ops.layer_norm_mod(x, sc, sh))
    gbs = 2 * x.numel() * 2 / dt / 1e9
    print(f"  layer_norm_mod: {dt*1e3:7.3f} ms  {gbs:7.0f} GB/s")
    dt = timeit(lambda: ops.gate_residual(x, sc, x))
    print(f"  gate_residual:  {dt*1e3:7.3f} ms  {3*x.numel()*2/dt/1e9:7.0f} GB/s")
    qh = torch.randn(B, 24, S, 128, device=dev, dtype=torch.bfloat16)
    w = torch.randn(128, device=dev, dtype=torch.bfloat16)
    dt = timeit(lambda: ops.rms_norm(qh, w))
    print(f"  rms_norm qk:    {dt*1e3:7.3f} ms  {2*qh.numel()*2/dt/1e9:7.0f} GB/s")
    from comfyui_parallelanything_amd.ops import reference as R
    cs = R.rope_freqs(torch.arange(S, device=dev), 128)
    dt = timeit(lambda: ops.rope_apply(qh, cs))
    print(f"  rope_apply:     {dt*1e3:7.3f} ms  {2*qh.numel()*2/dt/1e9:7.0f} GB/s")


if __name__ == "__main__":
    main()
