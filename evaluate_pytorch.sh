#!/bin/bash
# Evaluator launch (reference: src/evaluate_pytorch.sh).
exec python distributed_evaluator.py \
  --model-dir "${MODEL_DIR:-output/models/}" \
  --network "${NET:-ResNet18}" --dataset "${DATASET:-cifar10}" "$@"
