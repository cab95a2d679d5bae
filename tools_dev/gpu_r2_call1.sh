#!/bin/bash
# Round-2 call 1 (run ON the GPU box): clean rebuild from source, full
# GPU test suite (incl. the new kres/epilogue/fault-injection/random-
# column tests), short headline bench, epilogue bench leg.
set -x
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out
cd $R

# force a from-clean rebuild (no mtime cache; build() always rebuilds)
rm -f marlin_amd/libmarlin_gpu.so marlin_amd/.so_sha256
time python -c 'import __graft_entry__; __graft_entry__.build()' \
  > gpurun_out/r2_build.log 2>&1
sha256sum marlin_amd/libmarlin_gpu.so >> gpurun_out/r2_build.log

timeout 2100 python -m pytest tests -m gpu -x -q \
  > gpurun_out/r2_gputests.log 2>&1
echo "pytest rc=$?" >> gpurun_out/r2_gputests.log

python -c 'import __graft_entry__; __graft_entry__.smoke()' \
  > gpurun_out/r2_smoke.log 2>&1

# headline bench (driver-contract form, short)
timeout 500 python bench.py --steps 3 --warmup 1 \
  > gpurun_out/r2_bench20000.log 2>&1
# epilogue leg (config 5 fused (A*B)^T + D, device-resident)
timeout 500 python bench.py --workload epilogue --steps 3 --warmup 1 \
  --no-cpu-baseline > gpurun_out/r2_bench_epilogue.log 2>&1

tail -4 gpurun_out/r2_gputests.log
tail -2 gpurun_out/r2_smoke.log
tail -1 gpurun_out/r2_bench20000.log
tail -1 gpurun_out/r2_bench_epilogue.log
