#!/usr/bin/env bash
# 8-rank DDP example launch (reference examples/horovod/cluster.yaml analog).
set -euo pipefail
cd "$(dirname "$0")/../.."
N=${N:-8}
python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
  --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29501}" \
  examples/ddp/train_ddp.py "$@"
