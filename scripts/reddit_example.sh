#!/bin/bash
# Reddit 4-part GCN, all four modes (reference: scripts/example/reddit_*.sh)
cd "$(dirname "$0")/.."
set -e
for MODE in Vanilla AdaQP-q AdaQP-p AdaQP; do
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
    --master-addr 127.0.0.1 --master-port 29500 \
    main.py --dataset reddit --model_name gcn --mode $MODE
done
