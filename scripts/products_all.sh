#!/bin/bash
# ogbn-products 8-part sweep over {gcn,sage} x modes (reference: scripts/*_all.sh)
set -e
for MODEL in gcn sage; do
  for MODE in Vanilla AdaQP; do
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 --master-port 29500 \
      main.py --dataset ogbn-products --model_name $MODEL --mode $MODE
  done
done
