#!/usr/bin/env python3
"""torch.profiler attribution of one FLUX denoising step (GPU)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from comfyui_parallelanything_amd.models.registry import flux_inputs, make_flux

m = make_flux(dev="cuda", dtype=torch.bfloat16)
x, t, c, kw = flux_inputs(8, dev="cuda", dtype=torch.bfloat16)
with torch.no_grad():
    for _ in range(2):
        m(x, t, context=c, **kw)
    torch.cuda.synchronize()
    with torch.profiler.profile(
        activities=[torch.profiler.ProfilerActivity.CUDA,
                    torch.profiler.ProfilerActivity.CPU],
    ) as prof:
        m(x, t, context=c, **kw)
        torch.cuda.synchronize()
print(prof.key_averages().table(sort_by="self_cuda_time_total", row_limit=28))
