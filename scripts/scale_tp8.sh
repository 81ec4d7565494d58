#!/bin/bash
# Llama-3-70B TP=8 over xGMI — BASELINE config #3 launch path.
# One rank per GPU, RCCL collectives, fused one-shot allreduce+RMSNorm
# for decode (auto-enabled at tp>1). Run on an 8-GPU MI355X node:
#
#   bash scripts/scale_tp8.sh [STEPS] [WARMUP]
#
# The TP=1-per-GPU DP weak-scaling curve is the driver's own
# SCALE_rNN.json run (plain bench.py at N=1,2,4,8); this script is the
# 70B tensor-parallel point.
set -e
STEPS=${1:-200}
WARMUP=${2:-100}
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
    --master-addr 127.0.0.1 --master-port 29555 \
    bench.py --gpus 8 --tp 8 --model llama-3-70b \
    --max-num-seqs 512 --steps "$STEPS" --warmup "$WARMUP"
