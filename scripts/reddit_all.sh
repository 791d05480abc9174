#!/bin/bash
# Reddit sweep: {gcn,sage} x partition counts x {Vanilla,AdaQP}, then the
# comparison table (reference: scripts/reddit_all.sh + README.md:136-138).
# Override for quick runs: PARTS="2" MODES="Vanilla" EXTRA="--scale 0.01
# --num_epochs 3" scripts/reddit_all.sh
set -e
cd "$(dirname "$0")/.."
PARTS="${PARTS:-2 4}"
MODELS="${MODELS:-gcn sage}"
MODES="${MODES:-Vanilla AdaQP}"
PORT="${PORT:-29500}"
for P in $PARTS; do
  for MODEL in $MODELS; do
    for MODE in $MODES; do
      python -m torch.distributed.run --nnodes=1 --nproc-per-node "$P" \
        --master-addr 127.0.0.1 --master-port "$PORT" \
        main.py --dataset reddit --model_name "$MODEL" --mode "$MODE" \
        --assign_scheme adaptive $EXTRA
    done
  done
done
python tools/results_table.py --root "${EXP:-exp}" --dataset reddit
