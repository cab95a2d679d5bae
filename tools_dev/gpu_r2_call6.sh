#!/bin/bash
# Round-2 call 6: ASAN with explicit logging, aux kernel-trace (gemv
# kernel-time ground truth), soak cycle (full suite + smoke + benches).
set -x
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out
cd $R

# --- ASAN with alive-probes and per-process ASAN logs -----------------
ASAN_RT=$(find /opt/rocm/lib/llvm -name "libclang_rt.asan-x86_64.so" | head -1)
echo "ASAN_RT=$ASAN_RT" > gpurun_out/r2c6_asan.log
MARLIN_ASAN=1 python -c 'import __graft_entry__; __graft_entry__.build()' \
  >> gpurun_out/r2c6_asan.log 2>&1
LD_PRELOAD=$ASAN_RT python -c 'print("asan python alive")' \
  >> gpurun_out/r2c6_asan.log 2>&1
LD_PRELOAD=$ASAN_RT python -c 'import marlin_amd, marlin_amd.engine as E; E.lib(); print("asan lib loads")' \
  >> gpurun_out/r2c6_asan.log 2>&1
LD_PRELOAD=$ASAN_RT ASAN_OPTIONS="detect_leaks=0:log_path=$R/gpurun_out/r2c6_asan_proc" \
  timeout 600 python -m pytest tests/test_gpu_parity.py -p no:cacheprovider -q -m gpu \
  -k "golden or tile or summa_single or kres or error_paths or rccl or zero_pad or dgemv" \
  >> gpurun_out/r2c6_asan.log 2>&1
echo "asan pytest rc=$?" >> gpurun_out/r2c6_asan.log
ls gpurun_out/r2c6_asan_proc* >> gpurun_out/r2c6_asan.log 2>&1

# --- normal rebuild ----------------------------------------------------
python -c 'import __graft_entry__; __graft_entry__.build()' \
  > gpurun_out/r2c6_build.log 2>&1

# --- aux kernel-trace: true per-kernel times ---------------------------
cd /tmp && export TMPDIR=/tmp
rocprofv3 --kernel-trace --stats -d $R/gpurun_out/r2c6_auxtrace -- \
  python $R/tools_dev/bench_aux_device.py > $R/gpurun_out/r2c6_auxtrace.log 2>&1
cd $R
for db in $(find gpurun_out/r2c6_auxtrace -name "*.db"); do
  python tools_dev/rocpd_stats.py kernel $db
done > gpurun_out/r2c6_auxkernels.txt 2>&1

# --- soak cycle: full suite + smoke + default & epilogue benches -------
timeout 2100 python -m pytest tests -m gpu -q > gpurun_out/r2c6_suite.log 2>&1
echo "suite rc=$?" >> gpurun_out/r2c6_suite.log
python -c 'import __graft_entry__; __graft_entry__.smoke()' \
  > gpurun_out/r2c6_smoke.log 2>&1
timeout 500 python bench.py --steps 3 --warmup 1 --no-cpu-baseline \
  > gpurun_out/r2c6_bench.log 2>&1
timeout 500 python bench.py --workload epilogue --steps 2 --warmup 1 \
  --no-cpu-baseline > gpurun_out/r2c6_epi.log 2>&1

tail -8 gpurun_out/r2c6_asan.log
cat gpurun_out/r2c6_auxkernels.txt
tail -2 gpurun_out/r2c6_suite.log
tail -1 gpurun_out/r2c6_smoke.log
grep -h '"value"' gpurun_out/r2c6_bench.log gpurun_out/r2c6_epi.log
