#!/bin/bash
# One-shot GPU validation (single MI355X box) — designed for:
#   /usr/local/graft/bin/gpurun --timeout 2400 -- 'bash tools/validate_gpu.sh'
# Writes per-step logs under gpurun_out/validate/; prints a PASS/FAIL table.
# Budget: ~12-18 min typical; each step is individually timeout-bounded and
# the engines carry their own 60 s stall watchdogs, so a wedged engine costs
# ~1 min, not its step cap.
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

STEPS=(build gpu_tests smoke bench_flagship bench_rcv1 bench_epsilon \
       bench_spill bench_graph native_dist_w1 dist_probe dist2_py \
       dist2_native)

step build        600 python build_hip.py --force
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
# multi-rank RCCL first light: 2 ranks on this box's GPU(s). The probe tells
# us whether this RCCL build accepts same-device ranks on a 1-GPU box.
step dist_probe 180 python -m torch.distributed.run --standalone \
  --local-addr 127.0.0.1 --nproc-per-node 2 tools/rccl_probe2.py
if [ "${RES[dist_probe]}" = PASS ]; then
  step dist2_py 300 python -m torch.distributed.run --standalone \
    --local-addr 127.0.0.1 --nproc-per-node 2 bench.py --gpus 2 \
    --device cuda:0 --rows 400000 --steps 400 --warmup 100
  step dist2_native 300 env ASYNCAMD_DIST_ENGINE=native \
    python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
    --nproc-per-node 2 bench.py --gpus 2 --device cuda:0 --rows 400000 \
    --steps 400 --warmup 100
fi

echo "================ SUMMARY ================"
for k in "${STEPS[@]}"; do
  printf "%-16s %s\n" "$k" "${RES[$k]:-SKIPPED}"
done
grep -h '"metric"' "$OUT"/bench_*.log "$OUT"/dist2_*.log 2>/dev/null | head -8
