#!/bin/bash
# Round-2 call 7: ASAN retry (allocator_may_return_null for the HSA
# pool interceptor) + CPU-side ASAN fallback record + probe-sensitivity
# test + soak cycle.
set -x
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out
cd $R

ASAN_RT=$(find /opt/rocm/lib/llvm -name "libclang_rt.asan-x86_64.so" | head -1)
echo "ASAN_RT=$ASAN_RT" > gpurun_out/r2c7_asan.log
MARLIN_ASAN=1 python -c 'import __graft_entry__; __graft_entry__.build()' \
  >> gpurun_out/r2c7_asan.log 2>&1
# GPU attempt with the HSA-pool workaround
LD_PRELOAD=$ASAN_RT \
  ASAN_OPTIONS="detect_leaks=0:allocator_may_return_null=1:log_path=$R/gpurun_out/r2c7_asan_gpu" \
  timeout 600 python -m pytest tests/test_gpu_parity.py -p no:cacheprovider -q -m gpu \
  -k "golden or tile or summa_single or kres or error_paths or zero_pad or dgemv" \
  >> gpurun_out/r2c7_asan.log 2>&1
echo "asan GPU pytest rc=$?" >> gpurun_out/r2c7_asan.log
# CPU-side ASAN record (host paths of the instrumented .so: ABI surface,
# slab/plan helpers, oracle logic) — always meaningful
LD_PRELOAD=$ASAN_RT \
  ASAN_OPTIONS="detect_leaks=0:log_path=$R/gpurun_out/r2c7_asan_cpu" \
  timeout 600 python -m pytest tests/test_abi.py tests/test_summa_plan_all_grids.py \
  -p no:cacheprovider -q >> gpurun_out/r2c7_asan.log 2>&1
echo "asan CPU pytest rc=$?" >> gpurun_out/r2c7_asan.log

# normal rebuild + probe-sensitivity + soak
python -c 'import __graft_entry__; __graft_entry__.build()' \
  > gpurun_out/r2c7_build.log 2>&1
timeout 600 python -m pytest tests/test_gpu_parity.py -q -m gpu \
  -k "probe_detects or random_columns" > gpurun_out/r2c7_probe.log 2>&1
echo "probe rc=$?" >> gpurun_out/r2c7_probe.log
timeout 2100 python -m pytest tests -m gpu -q > gpurun_out/r2c7_suite.log 2>&1
echo "suite rc=$?" >> gpurun_out/r2c7_suite.log
timeout 500 python bench.py --steps 3 --warmup 1 --no-cpu-baseline \
  > gpurun_out/r2c7_bench.log 2>&1

grep -E "rc=|passed|failed|alive" gpurun_out/r2c7_asan.log
tail -2 gpurun_out/r2c7_probe.log
tail -2 gpurun_out/r2c7_suite.log
grep -h '"value"' gpurun_out/r2c7_bench.log
