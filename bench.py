#!/usr/bin/env python3
"""Flagship benchmark: FLUX.1-dev-class MMDiT, 1024x1024, batch 8, bf16.

Measures the reference's headline metric class (sec/it of the denoising
step; BASELINE.json: "sec/it + images/sec, FLUX.1-dev 1024^2 batch=8 at
1/2/4/8 MI355X") on synthetic latents and random-init weights.

One step == one full denoising iteration of the parallel engine:
scatter the latent batch over the GPUs (RCCL p2p over xGMI), run the
replicas' forward on their chunks, gather the noise predictions back to the
lead rank. Global batch is FIXED as N grows (strong scaling — the
reference's batch-21-over-2-GPUs headline has the same shape).

Launch (the driver's contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from comfyui_parallelanything_amd.models.registry import MODELS  # noqa: E402
from comfyui_parallelanything_amd.parallel.dist import (  # noqa: E402
    all_max,
    barrier,
    gatherv,
    init_distributed,
    scatterv,
)
from comfyui_parallelanything_amd.parallel.replicate import broadcast_module  # noqa: E402
from comfyui_parallelanything_amd.parallel.split import compute_split_sizes  # noqa: E402


def parse_args(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--model", default="flux",
                    choices=sorted(MODELS), help="model family")
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--px", type=int, default=1024)
    ap.add_argument("--dtype", default="bf16",
                    choices=["bf16", "fp16", "fp32", "fp8"],
                    help="fp8 = bf16 activations + e4m3fn GEMM weights (opt-in; NOT the headline dtype)")
    ap.add_argument("--tiny", action="store_true",
                    help="tiny config (CPU debugging only)")
    ap.add_argument("--no-graph", dest="graph", action="store_false",
                    help="disable hipGraph capture of the per-rank forward")
    ap.add_argument("--weights", default=None,
                    help="comma-separated per-rank split weights (e.g. 60,40 "
                         "for the Z-Image headline config); default even")
    ap.add_argument("--json-out", default=None)
    return ap.parse_args(argv)


DTYPES = {"bf16": torch.bfloat16, "fp16": torch.float16,
          "fp32": torch.float32, "fp8": torch.bfloat16}


def main(argv=None):
    args = parse_args(argv)
    on_gpu = torch.cuda.is_available()
    tiny = args.tiny or not on_gpu
    dtype = DTYPES[args.dtype] if (on_gpu or args.tiny) else torch.float32

    info = init_distributed()
    n = info.world_size if info.world_size > 1 else args.gpus
    if info.world_size == 1 and args.gpus > 1:
        raise SystemExit(
            "multi-GPU bench runs one process per GPU: launch via "
            "torch.distributed.run --nproc-per-node N (RCCL)."
        )
    dev = info.device

    if args.model in ("sd15", "sdxl") and on_gpu:
        # MIOpen conv algo search (cached in-process) for the UNet families
        torch.backends.cudnn.benchmark = True
    if on_gpu:
        # pre-tuned hipBLASLt algorithm selection (measured -1.0% flagship;
        # PA_NO_TUNABLEOP=1 disables)
        from comfyui_parallelanything_amd.utils.tunable import (
            enable_tuned_gemms,
        )

        enable_tuned_gemms()
    make, make_inputs = MODELS[args.model]
    model = make(dev=dev, dtype=dtype, tiny=tiny)
    # replicate(): rank0's weights to every replica, flat bucketed RCCL bcast
    broadcast_module(model, src_rank=0)
    if args.dtype == "fp8":
        from comfyui_parallelanything_amd.models.quant import quantize_fp8

        quantize_fp8(model)

    if args.weights:
        ws = [float(w) for w in args.weights.split(",")]
        if len(ws) != n:
            raise SystemExit("--weights count must equal the rank count")
        total_w = sum(ws)
        sizes = compute_split_sizes(args.batch, [w / total_w for w in ws])
    else:
        sizes = compute_split_sizes(args.batch, [1.0 / n] * n)
    if args.model.startswith("wan"):
        x, t, ctx, kw = make_inputs(args.batch, dev=dev, dtype=dtype, tiny=tiny)
    else:
        x, t, ctx, kw = make_inputs(args.batch, px=args.px, dev=dev, dtype=dtype,
                                    tiny=tiny)

    # static conditioning is scattered ONCE and stays resident per rank;
    # the latent x and the noise prediction move every step.
    my_ctx = scatterv(ctx, sizes, info, template=ctx)
    my_kw = {k: scatterv(v, sizes, info, template=v) for k, v in kw.items()}
    my_t_full = scatterv(t, sizes, info, template=t)

    # hipGraph capture of the per-rank forward: one graph replay per step
    # instead of ~2000 eager launches; RCCL scatter/gather stay eager
    # around it. Static I/O buffers are copied into/out of per step.
    graph_state = {}
    # hipGraph capture while an RCCL process group is live is UNPROVEN on
    # multi-rank hardware (this round's boxes lease one GPU): until a real
    # N>1 run validates it, multi-rank runs stay eager — launch overhead
    # measured at ~1% of a flux batch-8 step, a hang would cost the whole
    # scaling run. Override with PA_GRAPH_MULTIRANK=1 once proven.
    if (
        args.graph
        and info.world_size > 1
        and os.environ.get("PA_GRAPH_MULTIRANK") != "1"
    ):
        args.graph = False
        if info.is_lead:
            print("# world_size>1: hipGraph capture disabled "
                  "(set PA_GRAPH_MULTIRANK=1 to enable)", file=sys.stderr)

    @torch.no_grad()
    def run_model(my_x, my_t):
        if my_x.shape[0] == 0:
            # zero-size rank (more ranks than samples): nothing to compute,
            # output mirrors the latent's trailing shape
            return torch.empty_like(my_x)
        if not graph_state:
            return model(my_x, my_t, context=my_ctx, **my_kw)
        graph_state["x"].copy_(my_x)
        graph_state["t"].copy_(my_t)
        graph_state["graph"].replay()
        return graph_state["out"]

    @torch.no_grad()
    def capture_graph(my_x, my_t):
        sx = my_x.clone()
        st_ = my_t.clone()
        g = torch.cuda.CUDAGraph()
        torch.cuda.synchronize()
        with torch.cuda.graph(g):
            out = model(sx, st_, context=my_ctx, **my_kw)
        graph_state.update({"graph": g, "x": sx, "t": st_, "out": out})

    @torch.no_grad()
    def step(i: int) -> None:
        my_x = scatterv(x if info.is_lead else None, sizes, info, template=x)
        my_t = my_t_full * 0 + (1.0 - i / max(1, args.steps + args.warmup))
        eps = run_model(my_x, my_t)
        out = gatherv(eps, sizes, info, dst=0)
        if info.is_lead:
            # sampler update on the lead (Euler-style), keeps x live
            x.sub_(0.01 * out.to(x.dtype))

    for i in range(args.warmup):
        step(i)
        if i == 0 and args.graph and on_gpu and not graph_state:
            try:
                my_x = x[: sizes[info.rank]] if info.is_lead else torch.zeros(
                    (sizes[info.rank], *x.shape[1:]), dtype=x.dtype, device=dev
                )
                capture_graph(my_x, my_t_full)
            except Exception as err:  # noqa: BLE001
                print(f"# graph capture failed ({err!r}); running eager",
                      file=sys.stderr)
                graph_state.clear()

    barrier(info)
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(args.warmup + i)
    barrier(info)
    if on_gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    dt = all_max(dt, info)

    result = None
    if info.is_lead:
        sec_per_it = dt / args.steps
        images_per_s = args.batch * args.steps / dt
        result = {
            "metric": "images_per_s (whole-job; sec/it in ms_per_step), "
                      f"{args.model} {args.px}^2 batch={args.batch}",
            "value": round(images_per_s, 4),
            "unit": "images/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(sec_per_it * 1000.0, 2),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": args.dtype if on_gpu else "fp32",
            "data": "synthetic (random-init weights, random latents/context)",
            "config": {
                "model": {
                    "flux": "FLUX.1-dev-class MMDiT 12B",
                    "zimage": "Z-Image-class DiT 6B",
                    "sdxl": "SDXL-class UNet 2.6B",
                    "sd15": "SD1.5-class UNet",
                    "sd3": "SD3.5-Large-class MMDiT 8B",
                    "wan": "WAN2.2-class video DiT 14B",
                    "wan5b": "WAN2.2-class dense video DiT 5B",
                    "wan_i2v": "WAN2.2-class I2V video DiT 14B",
                }[args.model] + (" [TINY DEBUG CONFIG]" if tiny else ""),
                "global_batch": args.batch,
                "resolution": args.px,
                "seq_len": {
                    # tokens entering self-attention at the named config
                    "flux": (args.px // 16) ** 2 + 512,
                    "sd3": (args.px // 16) ** 2 + 154,
                    "zimage": (args.px // 16) ** 2 + 64,
                    "sdxl": (args.px // 16) ** 2,   # deepest-level grid
                    "sd15": (args.px // 16) ** 2,
                    "wan": 21 * 45 * 80,            # 720p x 21f, patch (1,2,2)
                    "wan5b": 21 * 45 * 80,
                    "wan_i2v": 21 * 45 * 80,
                }.get(args.model),
                "parallelism": f"dp{n}" + (
                    f" weighted {args.weights}" if args.weights else ""),
            },
        }
        # deterministic end-state fingerprint: lets the world-N gloo tests
        # assert N-rank scatter/forward/gather == single-process math
        result["x_checksum"] = round(float(x.float().abs().mean()), 8)
        line = json.dumps(result)
        print(line)
        if args.json_out:
            with open(args.json_out, "w") as f:
                f.write(line + "\n")
    return result


if __name__ == "__main__":
    main()
