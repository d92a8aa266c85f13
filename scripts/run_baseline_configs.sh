#!/bin/bash
# The five BASELINE.json configs (synthetic data, random-init weights).
# Single-GPU forms shown; multi-GPU via torch.distributed.run as printed.
set -x

# 1. SD1.5 256^2 batch=2, [cpu,cpu] 50/50 — plumbing, no GPU needed
python -m comfyui_parallelanything_amd.cli --model sd15 --devices cpu,cpu \
    --percent 50,50 --batch 2 --px 256 --steps 4 --no-balance

# 2. SDXL 1024^2 batch=4 on 2x MI355X 50/50 (UNet path)
python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 bench.py --gpus 2 --model sdxl --batch 4 \
    --px 1024 --steps 10 --warmup 3

# 3. Z_IMAGE Turbo 1024^2 batch=21 on 2x MI355X 60/40 (load-balancer path)
python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 bench.py --gpus 2 --model zimage --batch 21 \
    --px 1024 --weights 60,40 --steps 10 --warmup 3

# 4. FLUX.1-dev bf16 1024^2 batch=8 on 8x MI355X even split (flagship)
python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
    --master-addr 127.0.0.1 bench.py --gpus 8 --steps 10 --warmup 3

# 5. WAN2.2 720p batch=4 on 4x MI355X (video / temporal-attention path)
python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
    --master-addr 127.0.0.1 bench.py --gpus 4 --model wan --batch 4 \
    --steps 5 --warmup 2
