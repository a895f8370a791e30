#!/usr/bin/env bash
# Experiment sweep — the reference's run.sh grid (run.sh:25-53) on the
# MI355X framework: {dbs on/off} x {cifar10, cifar100} x four CV models,
# one-cycle LR enabled, fail-fast.  Each run is skipped automatically if
# its rank-0 log already exists (idempotent harness, dbs.py:528-534).
set -e

WS=${WS:-4}
BATCH=${BATCH:-512}
EPOCHS=${EPOCHS:-10}
GPUS=${GPUS:-0,1,2,3}

for DBS in true false; do
  for DS in cifar10 cifar100; do
    for MODEL in resnet densenet googlenet regnet; do
      echo "=== model=$MODEL ds=$DS dbs=$DBS ==="
      python dbs.py -d false -ws "$WS" -b "$BATCH" -e "$EPOCHS" \
        -ds "$DS" -m "$MODEL" -dbs "$DBS" -gpu "$GPUS" -ocp true
    done
  done
done
