#!/bin/bash
# LR grid-search harness (reference: src/tune.sh + tiny_tuning_parser.py):
# short runs over an lr grid, scraping the step-100 loss from the JSON logs.
set -e
STEPS=${STEPS:-100}
NET=${NET:-ResNet18}
CODE=${CODE:-svd}
OUT=${OUT:-tune_out}
mkdir -p "$OUT"
for e in 7 6 5 4 3 2 1; do
  LR=$(python -c "print(2**-$e)")
  echo "=== lr=$LR ==="
  python single_machine.py --network "$NET" --dataset cifar10 --code "$CODE" \
    --batch-size 8 --lr "$LR" --max-steps "$STEPS" --log-interval "$STEPS" \
    > "$OUT/lr_${LR}.log" 2>&1 || true
  python tiny_tuning_parser.py "$OUT/lr_${LR}.log" --lr "$LR"
done
