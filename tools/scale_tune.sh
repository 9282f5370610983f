#!/bin/bash
# scale_tune.sh — multi-GPU tuning sweep for round 2 (needs an 8-GPU node;
# this round's boxes are 1-GPU, so it has never run — first action of round
# 2 once a multi-GPU box is reachable).
#
# Sweeps, at 1024^3 Float64 x->y:
#   - process grids: the BASELINE 2x4 (pairwise exchange, 1 xGMI link/rank)
#     vs 4x2 (3 peers, 3 links) vs 8x1 (7 peers, 7 links — the per-hop
#     optimum by the link-bound model in DESIGN.md);
#   - exchange chunking: 1 (single grouped exchange, the shipped default)
#     vs 2/4/8 (PENCILHIP_EXCHANGE_CHUNKS overlap of unpack with in-flight chunks).
# Every line self-verifies (bench's exact checksum).

set -x
cd "$(dirname "$0")/.."
N=${1:-8}
STEPS=${2:-10}

run() {
  python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
    --master-addr 127.0.0.1 --master-port 29604 bench.py \
    --gpus "$N" --steps "$STEPS" --warmup 3 --no-cpu-baseline "$@" \
    2>/dev/null | tail -1
}

echo "=== grids (chunks=1) ==="
run
run --grid 4 2
run --grid 8 1
echo "=== chunking (default grid) ==="
run --chunks 2
run --chunks 4
run --chunks 8
echo "=== identity + double at the best grid (permuted is the default) ==="
run --identity
run --double
