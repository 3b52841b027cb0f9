#!/bin/bash
# Launch PS training on every visible GPU of this node (1 rank per GPU over
# RCCL).  Usage: tools/run_node.sh [distributed_nn.py args...]
set -e
export MIOPEN_FIND_MODE=${MIOPEN_FIND_MODE:-1}
cd "$(dirname "$0")/.."
NPROC=${NPROC:-$(python -c 'import torch;print(max(1,torch.cuda.device_count()))')}
exec python -m torch.distributed.run --nnodes="${NNODES:-1}" \
  --node-rank "${NODE_RANK:-0}" --nproc-per-node "$NPROC" \
  --master-addr "${MASTER_ADDR:-127.0.0.1}" --master-port "${MASTER_PORT:-29517}" \
  distributed_nn.py "$@"
