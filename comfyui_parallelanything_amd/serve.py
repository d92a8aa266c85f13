"""Minimal HTTP serving surface over the parallel engine.

The reference's serving story is "run inside ComfyUI"; headless users get
our CLI. This adds the third deployment shape — a FastAPI app exposing
the denoise loop over a parallel-installed model — so the engine can sit
behind a load balancer without a ComfyUI process. Conditioning is
bring-your-own (there is no text encoder in scope, matching the
reference, whose node also only touches the diffusion model): /generate
takes a seed and synthesizes latents/context of the model's native
shapes.

Run:  python -m comfyui_parallelanything_amd.serve \
          --model flux --devices cuda:0,cuda:1 --percent 60,40 --port 8188
"""
# NOTE: no `from __future__ import annotations` here — it stringifies the
# request-model annotations defined inside create_app(), and FastAPI then
# cannot resolve them (the body model silently degrades to a query param).
import threading
import time
from typing import List, Optional

import torch

from .models.registry import MODELS
from .parallel.chain import DeviceChain, make_entry
from .parallel.engine import ParallelEngine, install_parallel_forward
from .parallel.pipeline import configure_pipeline
from .sampling import SAMPLERS, flow_sigmas, karras_sigmas
from .utils.profiling import StepTimer


def create_app(
    model_name: str = "flux",
    devices: Optional[List[str]] = None,
    percents: Optional[List[float]] = None,
    dtype: torch.dtype = torch.bfloat16,
    tiny: bool = False,
    microbatches: int = 1,
    hip_graphs: bool = False,
    fp8: bool = False,
):
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    devices = devices or (["cuda:0"] if torch.cuda.is_available() else ["cpu"])
    percents = percents or [100.0 / len(devices)] * len(devices)
    if tiny or not torch.cuda.is_available():
        tiny, dtype = True, torch.float32

    if torch.cuda.is_available():
        from .utils.tunable import enable_tuned_gemms

        enable_tuned_gemms()
    chain = DeviceChain.from_list(
        [make_entry(d, p) for d, p in zip(devices, percents)]
    )
    make, make_inputs = MODELS[model_name]
    model = make(dev=chain.lead, dtype=dtype, tiny=tiny)
    if fp8:
        from .models.quant import quantize_fp8

        quantize_fp8(model)
    engine = ParallelEngine(chain, use_hip_graphs=hip_graphs)
    engine.setup(model)
    configure_pipeline(engine, microbatches=microbatches)
    install_parallel_forward(model, engine)
    timer = StepTimer(devices=list(chain.devices))

    class GenerateRequest(BaseModel):
        batch: int = 1
        steps: int = 8
        seed: int = 0
        sampler: str = "euler"
        shift: float = 1.0

    app = FastAPI(title="parallelanything-amd")
    app.state.engine = engine
    app.state.model = model
    # sync endpoints run in Starlette's threadpool: serialize engine use
    # (the per-device stream scheduler is single-caller by design)
    gen_lock = threading.Lock()

    @app.get("/healthz")
    def healthz():
        return {
            "status": "ok",
            "model": model_name,
            "devices": list(chain.devices),
            "weights": list(chain.weights),
            "dtype": str(dtype),
            "fp8": fp8,
        }

    @app.post("/generate")
    def generate(req: GenerateRequest):
        if req.sampler not in SAMPLERS:
            raise HTTPException(400, f"unknown sampler {req.sampler!r}; "
                                     f"have {sorted(SAMPLERS)}")
        if not 1 <= req.batch <= 64 or not 1 <= req.steps <= 200:
            raise HTTPException(400, "batch must be 1..64, steps 1..200")
        with gen_lock:
            torch.manual_seed(req.seed)
            x, _, c, kw = make_inputs(req.batch, dev=chain.lead, dtype=dtype,
                                      tiny=tiny)
            sig = (karras_sigmas(req.steps) if model_name in ("sd15", "sdxl")
                   else flow_sigmas(req.steps, shift=req.shift))
            timer.start()
            t0 = time.perf_counter()
            with torch.no_grad():
                out = SAMPLERS[req.sampler](model, x, sig, context=c, **kw)
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            timer.stop(images=req.batch)
        of = out.float()
        return {
            "shape": list(out.shape),
            "mean": of.mean().item(),
            "std": of.std().item(),
            "finite": bool(torch.isfinite(of).all().item()),
            "time_s": dt,
            "images_per_s": req.batch / dt,
        }

    @app.get("/stats")
    def stats():
        return timer.summary() if timer.steps else {"steps": 0}

    return app


def main(argv=None) -> None:
    import argparse

    import uvicorn

    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="flux", choices=sorted(MODELS))
    ap.add_argument("--devices", default=None)
    ap.add_argument("--percent", default=None)
    ap.add_argument("--port", type=int, default=8188)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--tiny", action="store_true")
    ap.add_argument("--microbatches", type=int, default=1)
    ap.add_argument("--hip-graphs", action="store_true")
    ap.add_argument("--fp8", action="store_true",
                    help="e4m3fn serving mode (quantize_fp8; GPU only)")
    args = ap.parse_args(argv)
    devices = args.devices.split(",") if args.devices else None
    percents = ([float(p) for p in args.percent.split(",")]
                if args.percent else None)
    app = create_app(args.model, devices, percents, tiny=args.tiny,
                     microbatches=args.microbatches,
                     hip_graphs=args.hip_graphs, fp8=args.fp8)
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
