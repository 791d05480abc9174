#!/bin/bash
# amazonProducts sweep: {gcn,sage} x partition counts x {Vanilla,AdaQP}
# + table (reference: scripts/amazon_all.sh). Largest graph: one MI355X
# holds it; 8-part needs the 8-GPU node. PARTS/MODES/MODELS/EXTRA env.
set -e
cd "$(dirname "$0")/.."
PARTS="${PARTS:-4 8}"
MODELS="${MODELS:-gcn sage}"
MODES="${MODES:-Vanilla AdaQP}"
PORT="${PORT:-29502}"
for P in $PARTS; do
  for MODEL in $MODELS; do
    for MODE in $MODES; do
      python -m torch.distributed.run --nnodes=1 --nproc-per-node "$P" \
        --master-addr 127.0.0.1 --master-port "$PORT" \
        main.py --dataset amazonProducts --model_name "$MODEL" --mode "$MODE" \
        --assign_scheme adaptive $EXTRA
    done
  done
done
python tools/results_table.py --root "${EXP:-exp}" --dataset amazonProducts
