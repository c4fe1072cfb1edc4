#!/usr/bin/env bash
# Launch the flagship benchmark on all 8 MI355X GPUs of one node
# (the analog of the reference's AWS cluster.yaml deployments: one rank per
# GPU over RCCL/xGMI instead of a Ray autoscaler cluster).
set -euo pipefail
cd "$(dirname "$0")/.."
N=${N:-8}
python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
  --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29500}" \
  bench.py --gpus "$N" "$@"
