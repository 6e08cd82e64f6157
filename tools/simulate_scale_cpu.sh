#!/bin/bash
# CPU (gloo) simulation of the round-end scale run's TOPOLOGY: the exact
# bench.py dist path at world = 2/4/8 with the flagship preset's worker
# pool, tiny data. Catches protocol/topology regressions without a GPU;
# absolute numbers are CPU-bound and only comparable to each other.
#   bash tools/simulate_scale_cpu.sh [python|native|both]
set -u
ENGINES=${1:-both}
PORT=29650
run() {  # run <world> <engine>
  local w=$1 eng=$2
  PORT=$((PORT + 1))
  local env=()
  [ "$eng" = native ] && env=(ASYNCAMD_DIST_ENGINE=native)
  local out
  out=$(env "${env[@]}" timeout 400 python -m torch.distributed.run \
        --nnodes=1 --nproc-per-node "$w" --master-addr 127.0.0.1 \
        --master-port $PORT bench.py --gpus "$w" --steps 100 --warmup 20 \
        --model asgd-mnist8m --rows 8000 --cols 64 2>/dev/null \
        | grep -o '"value": [0-9.]*' | grep -o '[0-9.]*')
  printf "%-8s world=%d  %10s updates/s\n" "$eng" "$w" "${out:-FAIL}"
}
for w in 2 4 8; do
  [ "$ENGINES" != native ] && run "$w" python
  [ "$ENGINES" != python ] && run "$w" native
done
true
