#!/usr/bin/env bash
# Convenience driver for a full measurement pass on an N-GPU MI355X node.
# Usage: tools/run_all_benches.sh [NGPUS]   (default: all visible GPUs)
set -e
cd "$(dirname "$0")/.."
N=${1:-$(python -c "import torch;print(torch.cuda.device_count())")}
RUN="python -m torch.distributed.run --nnodes=1 --master-addr 127.0.0.1"

echo "== flagship allreduce (symmetric default) =="
for n in 1 2 4 8; do
  [ "$n" -le "$N" ] || continue
  if [ "$n" -eq 1 ]; then python bench.py --gpus 1
  else $RUN --nproc-per-node $n --master-port 29801 bench.py --gpus $n; fi
done

echo "== allreduce sweep (8B..1GB bf16) =="
$RUN --nproc-per-node $N --master-port 29803 bench_sweep.py || true

echo "== uccl vs RCCL side-by-side =="
[ "$N" -ge 2 ] && $RUN --nproc-per-node $N --master-port 29805 bench_compare.py || true

echo "== EP dispatch/combine p50 (BASELINE shape) =="
$RUN --nproc-per-node $N --master-port 29807 bench_ep.py || true

echo "== P2P bandwidth =="
python bench_p2p.py || true
