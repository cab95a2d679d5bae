#!/bin/bash
# Round-2 call 5: gemv-fix validation, ASAN record (fixed build), one
# more paired phase check, driver-form bench + CLI.
set -x
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out
cd $R

# ASAN leg (build now skips CLI/JNI links)
ASAN_RT=$(find /opt/rocm/lib/llvm -name "libclang_rt.asan-x86_64.so" | head -1)
echo "ASAN_RT=$ASAN_RT" > gpurun_out/r2c5_asan.log
MARLIN_ASAN=1 python -c 'import __graft_entry__; __graft_entry__.build()' \
  >> gpurun_out/r2c5_asan.log 2>&1
LD_PRELOAD=$ASAN_RT ASAN_OPTIONS=detect_leaks=0 \
  timeout 600 python -m pytest tests/test_gpu_parity.py -q -m gpu \
  -k "golden or tile or summa_single or kres or error_paths or rccl or zero_pad or dgemv" \
  >> gpurun_out/r2c5_asan.log 2>&1
echo "asan pytest rc=$?" >> gpurun_out/r2c5_asan.log

# normal rebuild
python -c 'import __graft_entry__; __graft_entry__.build()' \
  > gpurun_out/r2c5_build.log 2>&1

# gemv bandwidth after the 4-acc unroll + parity
timeout 300 python tools_dev/bench_aux_device.py > gpurun_out/r2c5_aux.log 2>&1
timeout 600 python -m pytest tests/test_gpu_parity.py tests/test_gpu_elementwise.py \
  -q -m gpu -k "gemv or elementwise or transpose or sum" \
  > gpurun_out/r2c5_gemvtests.log 2>&1
echo "rc=$?" >> gpurun_out/r2c5_gemvtests.log

# paired phase check #2 (same box)
B="python bench.py --gpus 1 --warmup 1 --no-cpu-baseline"
timeout 400 $B --steps 3 > gpurun_out/r2c5_f64_base.log 2>&1
MARLIN_GEMM_PHASE=8 timeout 400 $B --steps 3 > gpurun_out/r2c5_f64_ph8.log 2>&1

# driver-form default bench (incl. cpu_baseline) + CLI verify
timeout 500 python bench.py > gpurun_out/r2c5_bench_default.log 2>&1
./marlinx verify 300 200 100 > gpurun_out/r2c5_cli.log 2>&1
./marlinx bench 8192 8192 8192 2 1 >> gpurun_out/r2c5_cli.log 2>&1

tail -4 gpurun_out/r2c5_asan.log
head -4 gpurun_out/r2c5_aux.log
tail -2 gpurun_out/r2c5_gemvtests.log
grep -h '"value"' gpurun_out/r2c5_f64_base.log gpurun_out/r2c5_f64_ph8.log gpurun_out/r2c5_bench_default.log
tail -2 gpurun_out/r2c5_cli.log
