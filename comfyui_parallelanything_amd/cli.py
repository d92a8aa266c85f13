"""Headless CLI: run the parallel engine without ComfyUI in the loop.

The reference's only config surface is ComfyUI node inputs (SURVEY.md §5
Config); this CLI is the headless equivalent for benchmarks and serving:

  python -m comfyui_parallelanything_amd.cli \
      --model flux --devices cuda:0,cuda:1 --percent 60,40 \
      --batch 8 --px 1024 --steps 20

Builds the DEVICE_CHAIN exactly as the nodes would, replicates, installs the
parallel forward, and runs a denoising loop with per-step metrics.
"""
from __future__ import annotations

import argparse
import json

import torch

from .models.registry import MODELS
from .parallel.chain import DeviceChain, make_entry
from .parallel.cleanup import cleanup_parallel_model
from .parallel.engine import ParallelEngine, install_parallel_forward
from .parallel.pipeline import configure_pipeline
from .utils.profiling import StepTimer, trace_range


def main(argv=None):
    ap = argparse.ArgumentParser(description="ParallelAnything headless runner")
    ap.add_argument("--model", default="flux", choices=sorted(MODELS))
    ap.add_argument("--devices", default="cpu,cpu",
                    help="comma-separated chain, e.g. cuda:0,cuda:1")
    ap.add_argument("--percent", default=None,
                    help="comma-separated percentages (default: even)")
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--px", type=int, default=1024)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--dtype", default="bf16",
                    choices=["bf16", "fp16", "fp32", "fp8"],
                    help="fp8 = e4m3fn serving mode (GPU chains only)")
    ap.add_argument("--tiny", action="store_true")
    ap.add_argument("--no-split", dest="split", action="store_false")
    ap.add_argument("--no-balance", dest="balance", action="store_false")
    ap.add_argument("--microbatches", type=int, default=1,
                    help="pipeline-mode micro-batches (>1 overlaps stages)")
    ap.add_argument("--hip-graphs", action="store_true",
                    help="capture repeated same-shape forwards into hipGraphs")
    ap.add_argument("--lora", default=None, metavar="PATH[:SCALE]",
                    help="merge a LoRA .safetensors into the weights before "
                         "replication (PEFT or kohya key conventions)")
    ap.add_argument("--json-out", default=None,
                    help="write the run summary as one JSON record")
    args = ap.parse_args(argv)

    devices = args.devices.split(",")
    pcts = (
        [float(p) for p in args.percent.split(",")]
        if args.percent
        else [100.0 / len(devices)] * len(devices)
    )
    if len(pcts) != len(devices):
        raise SystemExit("--percent count must match --devices count")
    dtype = {"bf16": torch.bfloat16, "fp16": torch.float16,
             "fp32": torch.float32, "fp8": torch.bfloat16}[args.dtype]
    tiny = args.tiny or not torch.cuda.is_available()
    if tiny and dtype != torch.float32:
        dtype = torch.float32

    if torch.cuda.is_available():
        from .utils.tunable import enable_tuned_gemms

        enable_tuned_gemms()
    chain = DeviceChain.from_list(
        [make_entry(d, p) for d, p in zip(devices, pcts)]
    )
    make, make_inputs = MODELS[args.model]
    model = make(dev=chain.lead, dtype=dtype, tiny=tiny)

    if args.lora:
        from .models.lora import merge_lora_file

        path, _, s = args.lora.partition(":")
        n = merge_lora_file(model, path, scale=float(s) if s else 1.0)
        print(f"[lora] merged {n} modules from {path}")

    if args.dtype == "fp8":
        from .models.quant import quantize_fp8

        quantize_fp8(model)

    engine = ParallelEngine(chain, workload_split=args.split,
                            auto_vram_balance=args.balance,
                            use_hip_graphs=args.hip_graphs)
    engine.setup(model)
    configure_pipeline(engine, microbatches=args.microbatches)
    install_parallel_forward(model, engine)

    if args.model.startswith("wan"):
        x, t, ctx, kw = make_inputs(args.batch, dev=chain.lead, dtype=dtype,
                                    tiny=tiny)
    else:
        x, t, ctx, kw = make_inputs(args.batch, px=args.px, dev=chain.lead,
                                    dtype=dtype, tiny=tiny)

    timer = StepTimer(devices=list(dict.fromkeys(chain.devices)))
    with torch.no_grad():
        for i in range(args.steps):
            timer.start()
            with trace_range(f"pa::denoise_step_{i}"):
                t_i = t * 0 + (1.0 - i / max(1, args.steps))
                eps = model(x, t_i, context=ctx, **kw)
                x = x - 0.01 * eps.to(x.dtype)
            if torch.device(chain.lead).type == "cuda":
                torch.cuda.synchronize(torch.device(chain.lead))
            s = timer.stop(args.batch)
            print(f"step {i:3d}: {s.wall_s*1000:8.1f} ms  "
                  f"{s.images_per_s:8.2f} img/s")
    summary = timer.summary()
    summary["config"] = {
        "model": args.model, "batch": args.batch, "px": args.px,
        "devices": devices, "percent": pcts,
        "dtype": str(dtype).replace("torch.", ""), "tiny": tiny,
        "engine": "in-process",
    }
    print(json.dumps(summary, indent=2))
    if args.json_out:
        with open(args.json_out, "w") as f:
            json.dump(summary, f)
            f.write("\n")
    cleanup_parallel_model(model)


if __name__ == "__main__":
    main()
