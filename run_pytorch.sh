#!/bin/bash
# Canonical launch (reference: src/run_pytorch.sh — mpirun -n 3 replaced by
# torchrun, 1 rank per GPU over RCCL; CPU/gloo when no GPU is visible).
# ResNet-18 / CIFAR-10 shape, bs 128, lr 0.01, SVD rank 3.
set -e
export MIOPEN_FIND_MODE=${MIOPEN_FIND_MODE:-1}
NPROC=${NPROC:-$(python -c 'import torch;print(max(1,torch.cuda.device_count()))')}
exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NPROC" \
  --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29517}" \
  distributed_nn.py \
  --network ResNet18 --dataset cifar10 --code svd --svd-rank 3 \
  --batch-size 128 --lr 0.01 --lr-shrinkage 0.95 --momentum 0.9 \
  --max-steps "${MAX_STEPS:-1000}" --log-interval 10 "$@"
