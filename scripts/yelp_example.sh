#!/bin/bash
# Yelp 4-part GraphSAGE (multilabel micro-F1), Vanilla vs AdaQP
cd "$(dirname "$0")/.."
set -e
for MODE in Vanilla AdaQP; do
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
    --master-addr 127.0.0.1 --master-port 29500 \
    main.py --dataset yelp --model_name sage --mode $MODE
done
