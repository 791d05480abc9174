#!/bin/bash
# AmazonProducts 8-part GCN (largest graph), AdaQP adaptive
cd "$(dirname "$0")/.."
set -e
python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
  --master-addr 127.0.0.1 --master-port 29500 \
  main.py --dataset amazonProducts --model_name gcn --mode AdaQP \
  --assign_scheme adaptive
