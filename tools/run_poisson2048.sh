#!/bin/bash
# BASELINE config 5 push-button (SURVEY §7 stage 5): 7-pt 3D Poisson
# 2048^3 (~8.6B rows, fp64) across the 8 MI355X of one node.
#
#   NGPUS=8 ./tools/run_poisson2048.sh              # assembled operator
#   NGPUS=8 ./tools/run_poisson2048.sh --matfree    # matrix-free (~81 GiB/rank)
#
# bench.py runs a pre-flight memory assertion BEFORE any allocation,
# calibrated on single-GPU slab measurements (tools/slab_probe.py,
# profiles/RESULTS.md): rank 0/8 assembled 157.2 GiB, matfree 80.9 GiB
# of the 288 GiB HBM3E per GPU.  A 2048^3 run that cannot fit fails in
# seconds with a clear message instead of OOMing the node.
set -euo pipefail
cd "$(dirname "$0")/.."
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}
GRID=${GRID:-2048}
STEPS=${STEPS:-30}
WARMUP=${WARMUP:-5}
NGPUS=${NGPUS:-8}
exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NGPUS" \
  --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29571}" \
  bench.py --gpus "$NGPUS" --config poisson7 --grid "$GRID" \
  --steps "$STEPS" --warmup "$WARMUP" "$@"
