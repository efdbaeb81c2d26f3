#!/usr/bin/env bash
# SCALE dry run: the exact invocation the driver uses for the weak-scaling
# curve (N=1,2,4,8 on one node, one rank per MI355X over RCCL/xGMI).
#
# Run on an 8-GPU node:   bash scripts/scale_dryrun.sh
# Single-GPU smoke:       bash scripts/scale_dryrun.sh 1
#
# Notes
#  * MASTER_ADDR is pinned to 127.0.0.1 (container hostnames may not
#    resolve); the port is rotated per N to avoid TIME_WAIT collisions.
#  * HSA_ENABLE_IPC_MODE_LEGACY=0 must stay exported (dmabuf IPC is the
#    only mode the host driver supports; RCCL fails without it).
#  * FAM_NCCL_DEBUG=1 turns on NCCL_DEBUG=WARN inside bench.py, so RCCL
#    ring/xGMI setup problems are printed instead of silently hanging;
#    FAM_PG_TIMEOUT bounds the rendezvous (default 600 s).
#  * Weak scaling: per-GPU population is fixed; the reported value is the
#    whole-job aggregate, so ideal scaling multiplies it by N.
set -euo pipefail
cd "$(dirname "$0")/.."

export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}
export FAM_NCCL_DEBUG=${FAM_NCCL_DEBUG:-1}
export FAM_PG_TIMEOUT=${FAM_PG_TIMEOUT:-120}

STEPS=${STEPS:-20}
WARMUP=${WARMUP:-5}
NS=${1:-"1 2 4 8"}

for N in $NS; do
  echo "=== N=$N ==="
  if [ "$N" = 1 ]; then
    python bench.py --gpus 1 --steps "$STEPS" --warmup "$WARMUP"
  else
    python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
      --master-addr 127.0.0.1 --master-port $((29500 + N)) \
      bench.py --gpus "$N" --steps "$STEPS" --warmup "$WARMUP"
  fi
done
