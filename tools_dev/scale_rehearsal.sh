#!/bin/bash
# Multi-GPU SUMMA rehearsal (VERDICT r01 item 1 pre-positioning).
# Run on ANY node with N MI355X GPUs (driver round-end, or by hand):
#   bash tools_dev/scale_rehearsal.sh [max_gpus]
# Produces gpurun_out/scale_rehearsal/*.log with, per N in 1,2,4,..max:
#   - the headline 20000^3 fp64 bench line (JSON),
#   - per-rank comm/GEMM overlap stats (MARLIN_SUMMA_DEBUG),
#   - at N=8: the MARLIN_GRID 4x2 vs 2x4 sweep and MARLIN_SUMMA_KB
#     2048/4096/8192 sweep,
#   - the config-4 k-resident leg (50000x4096x50000, zero broadcasts).
# Defaults chosen by analysis: grid 4x2, KB 4096 — at
# 20000^3/8GPU each panel's GEMM (~6 ms) covers its broadcasts (~2-3 ms
# on the comm stream), so comm should fully hide; the sweep verifies.
set -x
R=${GRAFT_REPO_ROOT:-$(cd "$(dirname "$0")/.." && pwd)}
cd $R
OUT=$R/gpurun_out/scale_rehearsal
mkdir -p $OUT
MAX=${1:-$(rocm-smi --showid 2>/dev/null | grep -c "^GPU" || echo 1)}
export HSA_ENABLE_IPC_MODE_LEGACY=0
export MASTER_ADDR=127.0.0.1

run() { # run N steps warmup extra_env... -- extra_args...
  local N=$1 TAG=$2; shift 2
  if [ "$N" = 1 ]; then
    timeout 600 python bench.py --gpus 1 --steps 3 --warmup 1 \
      --no-cpu-baseline "$@" > $OUT/$TAG.log 2>&1
  else
    timeout 900 python -m torch.distributed.run --nnodes=1 \
      --nproc-per-node $N --master-addr 127.0.0.1 \
      --master-port $((20000 + RANDOM % 20000)) \
      bench.py --gpus $N --steps 3 --warmup 1 --no-cpu-baseline "$@" \
      > $OUT/$TAG.log 2>&1
  fi
  grep -h '"metric"' $OUT/$TAG.log | tail -1
}

export MARLIN_SUMMA_DEBUG=1
for N in 1 2 4 8; do
  [ "$N" -le "$MAX" ] || continue
  run $N n${N}_default
done

if [ "$MAX" -ge 8 ]; then
  MARLIN_GRID=2x4 run 8 n8_grid2x4
  MARLIN_SUMMA_KB=2048 run 8 n8_kb2048
  MARLIN_SUMMA_KB=8192 run 8 n8_kb8192
  # bulk mode: panels = whole owner slabs (the all-gather-equivalent
  # fallback SURVEY §8e names — all comm up front, no pipelining)
  MARLIN_SUMMA_KB=20000 run 8 n8_kb_bulk
  # config 4: k-resident layout (zero steady-state xGMI traffic)
  MARLIN_BENCH_M=50000 MARLIN_BENCH_K=4096 MARLIN_BENCH_NN=50000 \
    run 8 n8_config4_kres
  # config 5: fp32 40000^2
  MARLIN_BENCH_DTYPE=f32 MARLIN_BENCH_N=40000 run 8 n8_config5_fp32
fi

grep -h '"metric"' $OUT/*.log | tail -20
grep -h "layout\|comm " $OUT/*.log | tail -40
