#!/bin/bash
# ogbn-products 8-part GraphSAGE, Vanilla then AdaQP — the BASELINE.json
# headline config (reference: scripts/example/products_{vanilla,adaqp}.sh)
set -e
cd "$(dirname "$0")/.."
for MODE in Vanilla AdaQP; do
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
    --master-addr 127.0.0.1 --master-port 29505 \
    main.py --dataset ogbn-products --model_name sage --mode $MODE \
    --assign_scheme adaptive $EXTRA
done
