#!/bin/bash
# Round-1 final evidence bundle (run ON the GPU box).
set -x
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out
cd $R

# C++ host CLI (marlinx) — verify + a small bench through the C ABI
./marlinx verify 300 200 100 > gpurun_out/marlinx.log 2>&1
./marlinx bench 8192 8192 8192 2 1 >> gpurun_out/marlinx.log 2>&1

# the driver-contract default bench (incl. cpu_baseline leg)
timeout 500 python bench.py > gpurun_out/bench_final.log 2>&1

cd /tmp && export TMPDIR=/tmp
B20="python $R/bench.py --gpus 1 --warmup 1 --no-cpu-baseline"
# kernel-trace stats (rocpd db)
rocprofv3 --kernel-trace --stats -d $R/gpurun_out/prof_final -- \
  $B20 --steps 2 > $R/gpurun_out/prof_final.log 2>&1
# FETCH_SIZE pass (own pass per the PMC rules)
rocprofv3 --pmc FETCH_SIZE -d $R/gpurun_out/pmc_fetch3 -o fetch3 \
  --output-format csv -- $B20 --steps 1 > $R/gpurun_out/pmc_fetch3.log 2>&1
# SQ pass (final kernel)
rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
  SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE SQ_ACTIVE_INST_ANY \
  -d $R/gpurun_out/pmc_sq3 -o sq3 --output-format csv -- \
  $B20 --steps 1 > $R/gpurun_out/pmc_sq3.log 2>&1

tail -3 $R/gpurun_out/marlinx.log
tail -1 $R/gpurun_out/bench_final.log
