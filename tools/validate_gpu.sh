#!/bin/bash
# One-shot GPU validation (single MI355X box) — designed for:
#   /usr/local/graft/bin/gpurun --timeout 2000 -- 'bash tools/validate_gpu.sh'
# Writes per-step logs under gpurun_out/validate/; prints a PASS/FAIL table.
# Budget: ~10-20 min typical; each step is individually timeout-bounded
# (worst case ~60 min if everything hangs to its limit — set gpurun
# --timeout accordingly, e.g. 2400).
set -u
OUT=gpurun_out/validate
mkdir -p "$OUT"
declare -A RES

step() {  # step <name> <timeout_s> <cmd...>
  local name=$1 tmo=$2; shift 2
  echo "=== $name ==="
  if timeout "$tmo" "$@" > "$OUT/$name.log" 2>&1; then
    RES[$name]=PASS
  else
    RES[$name]="FAIL($?)"
  fi
  tail -3 "$OUT/$name.log"
}

step build        600 python build_hip.py
step gpu_tests    900 python -m pytest tests -m gpu -x -q
step smoke        300 python __graft_entry__.py smoke
# all five BASELINE configs, short
step bench_flagship 300 python bench.py --model asgd-mnist8m --steps 2000 --warmup 300
step bench_rcv1     300 python bench.py --model asaga-rcv1 --steps 1500 --warmup 200
step bench_epsilon  300 python bench.py --model asgd-epsilon-delay --steps 1500 --warmup 200
step bench_spill    300 python bench.py --model asaga-mnist8m-hostspill --steps 400 --warmup 50
step bench_graph    300 python bench.py --model asgd-mnist8m --engine graph --steps 2000 --warmup 300
# C++ dist server, world=1 on the single GPU (channel-free path)
step native_dist_w1 300 python -m pytest tests/test_dist_native.py::test_native_dist_world1_gpu -q

echo "================ SUMMARY ================"
for k in build gpu_tests smoke bench_flagship bench_rcv1 bench_epsilon \
         bench_spill bench_graph native_dist_w1; do
  printf "%-16s %s\n" "$k" "${RES[$k]:-SKIPPED}"
done
grep -h '"metric"' "$OUT"/bench_*.log 2>/dev/null | head -8
