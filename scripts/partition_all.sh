#!/bin/bash
# offline partitioning for every dataset (reference: scripts/partition/*.sh)
cd "$(dirname "$0")/.."
set -e
for DS in reddit yelp ogbn-products amazonProducts; do
  python graph_partition.py --dataset $DS --partition_size ${1:-8}
done
