#!/bin/bash
# Multi-GPU xGMI bring-up dry run: executable
# UNMODIFIED on any >=2-GPU MI355X lease. Runs the peer-HBM visibility
# litmus, then the flagship bench at each GPU count.
set -e
cd "$(dirname "$0")/.."
export HSA_ENABLE_IPC_MODE_LEGACY=0
N=${1:-$(python -c 'import torch; print(torch.cuda.device_count())')}
echo "== xGMI litmus at world=$N =="
python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
    --master-addr 127.0.0.1 --master-port 29611 tools/xgmi_litmus.py
for n in 2 4 8; do
  [ "$n" -gt "$N" ] && break
  echo "== bench.py --gpus $n =="
  python -m torch.distributed.run --nnodes=1 --nproc-per-node "$n" \
      --master-addr 127.0.0.1 --master-port 29612 \
      bench.py --gpus "$n" --steps 20 --warmup 5
done
echo "XGMI DRY RUN COMPLETE"
