#!/bin/bash
# The five BASELINE.json configs through ONE harness (synthetic data,
# random-init weights). Each config is one command producing one JSON
# artifact in $OUT_DIR; n_gpus inside each record labels what actually ran.
# Requested GPU counts degrade to the GPUs present (this round's boxes
# lease ONE MI355X; the driver's round-end SCALE run covers N>1).
#
#   OUT_DIR=gpurun_out/baseline STEPS=10 bash scripts/run_baseline_configs.sh
set -e

OUT_DIR=${OUT_DIR:-gpurun_out/baseline}
STEPS=${STEPS:-10}
WARMUP=${WARMUP:-3}
mkdir -p "$OUT_DIR"
NGPU=$(python -c "import torch; print(torch.cuda.device_count())")
echo "# GPUs visible: $NGPU"

run_bench() {
    local name=$1 req=$2
    shift 2
    local n=$(( req < NGPU ? req : NGPU ))
    [ "$n" -lt 1 ] && n=1
    echo "### $name (requested ${req} GPUs, running on ${n})"
    if [ "$n" -gt 1 ]; then
        python -m torch.distributed.run --nnodes=1 --nproc-per-node "$n" \
            --master-addr 127.0.0.1 bench.py --gpus "$n" \
            --steps "$STEPS" --warmup "$WARMUP" "$@" \
            --json-out "$OUT_DIR/$name.json"
    else
        python bench.py --gpus 1 --steps "$STEPS" --warmup "$WARMUP" "$@" \
            --json-out "$OUT_DIR/$name.json"
    fi
}

# 1. SD1.5 256^2 batch=2, [cpu,cpu] 50/50 — plumbing, in-process engine,
#    no GPU needed (BASELINE config 1)
python -m comfyui_parallelanything_amd.cli --model sd15 --devices cpu,cpu \
    --percent 50,50 --batch 2 --px 256 --steps 4 --no-balance \
    --json-out "$OUT_DIR/config1_sd15_cpu.json"

# 2. SDXL 1024^2 batch=4 on 2x MI355X 50/50 even split (UNet path)
run_bench config2_sdxl 2 --model sdxl --batch 4 --px 1024

# 3. Z_IMAGE Turbo 1024^2 batch=21 on 2x MI355X 60/40 weighted split
#    (README headline, load-balancer path)
if [ "$NGPU" -ge 2 ]; then
    run_bench config3_zimage 2 --model zimage --batch 21 --px 1024 \
        --weights 60,40
else
    run_bench config3_zimage 1 --model zimage --batch 21 --px 1024
fi

# 4. FLUX.1-dev bf16 1024^2 batch=8 on 8x MI355X even split (flagship)
run_bench config4_flux 8 --model flux --batch 8 --px 1024

# 5. WAN2.2 I2V 720p batch=4 on 4x MI355X (video / temporal-attn path)
# (~45 s/step on ONE GPU at this 4x75600-token config: fewer steps)
STEPS=2 WARMUP=1 run_bench config5_wan_i2v 4 --model wan_i2v --batch 4

echo "### artifacts:"
ls -l "$OUT_DIR"
