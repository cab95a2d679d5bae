#!/bin/bash
# Round-2 call 2: validate epilogue rewrite + RCCL probe fix; run the
# k-phase and band-width experiments.
set -x
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out
cd $R

python -c 'import __graft_entry__; __graft_entry__.build()' \
  > gpurun_out/r2c2_build.log 2>&1

timeout 900 python -m pytest tests/test_gpu_parity.py -q -m gpu \
  -k "epilogue or rccl or vendor or zero_pad or kres or random_columns or transpose" \
  > gpurun_out/r2c2_tests.log 2>&1
echo "pytest rc=$?" >> gpurun_out/r2c2_tests.log

B="python bench.py --gpus 1 --warmup 1 --no-cpu-baseline"
# epilogue leg after the LDS-staged store rewrite
timeout 400 $B --workload epilogue --steps 3 > gpurun_out/r2c2_epi.log 2>&1
# fp64 20000^3: baseline vs k-phase stagger
timeout 400 $B --steps 3 > gpurun_out/r2c2_f64_base.log 2>&1
MARLIN_GEMM_PHASE=1 timeout 400 $B --steps 3 > gpurun_out/r2c2_f64_ph1.log 2>&1
# fp32 40000^2 band sweep (tune config 5's plain-gemm size)
MARLIN_BENCH_DTYPE=f32 MARLIN_BENCH_N=40000 timeout 400 $B --steps 2 \
  > gpurun_out/r2c2_f32_b8.log 2>&1
MARLIN_GEMM_BAND=16 MARLIN_BENCH_DTYPE=f32 MARLIN_BENCH_N=40000 \
  timeout 400 $B --steps 2 > gpurun_out/r2c2_f32_b16.log 2>&1
MARLIN_GEMM_BAND=4 MARLIN_BENCH_DTYPE=f32 MARLIN_BENCH_N=40000 \
  timeout 400 $B --steps 2 > gpurun_out/r2c2_f32_b4.log 2>&1

tail -3 gpurun_out/r2c2_tests.log
grep -h '"value"' gpurun_out/r2c2_*.log
