"""Eager vs hipGraph A/B for the in-process engine (lead-only, cuda:0).

Evidence run for engine.use_hip_graphs (see parallel/hipgraphs.py).
"""
import sys

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))

import time

import torch

from comfyui_parallelanything_amd.models.registry import flux_inputs, make_flux
from comfyui_parallelanything_amd.parallel.chain import DeviceChain, make_entry
from comfyui_parallelanything_amd.parallel.engine import ParallelEngine


def run(graphs: bool, tiny: bool, batch: int, iters: int) -> float:
    m = make_flux(dev="cuda:0", dtype=torch.bfloat16, tiny=tiny)
    eng = ParallelEngine(
        DeviceChain.from_list([make_entry("cuda:0", 100)]),
        auto_vram_balance=False,
        use_hip_graphs=graphs,
    )
    eng.setup(m)
    x, t, c, kw = flux_inputs(batch, dev="cuda:0", dtype=torch.bfloat16,
                              tiny=tiny)
    for _ in range(5):
        eng.forward(x, t, context=c, **kw)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        eng.forward(x, t, context=c, **kw)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters * 1000
    eng.release()
    del m
    torch.cuda.empty_cache()
    return dt


if __name__ == "__main__":
    for tiny, batch, iters in ((True, 2, 50), (False, 8, 15)):
        e = run(False, tiny, batch, iters)
        g = run(True, tiny, batch, iters)
        print(
            f"tiny={tiny} batch={batch}: eager {e:.2f} ms  "
            f"graph {g:.2f} ms  speedup {e / g:.3f}x"
        )
