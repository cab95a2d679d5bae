#!/bin/bash
# Round-2 call 8: soak (full suite + smoke) + endurance + 104 GiB
# big-GEMM revalidation + config-2 bench line.
set -x
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out
cd $R
python -c 'import __graft_entry__; __graft_entry__.build()' > gpurun_out/r2c8_build.log 2>&1
timeout 2100 python -m pytest tests -m gpu -q > gpurun_out/r2c8_suite.log 2>&1
echo "suite rc=$?" >> gpurun_out/r2c8_suite.log
python -c 'import __graft_entry__; __graft_entry__.smoke()' > gpurun_out/r2c8_smoke.log 2>&1
timeout 900 python tools_dev/endurance.py > gpurun_out/r2c8_endurance.log 2>&1
timeout 900 python tools_dev/big_gemm.py > gpurun_out/r2c8_biggemm.log 2>&1
B="python bench.py --gpus 1 --warmup 1 --no-cpu-baseline"
timeout 400 $B --steps 3 > gpurun_out/r2c8_bench20000.log 2>&1
timeout 400 $B --steps 5 --n 10000 > gpurun_out/r2c8_bench10000.log 2>&1
timeout 400 $B --steps 2 --m 50000 --k 4096 --nn 50000 > gpurun_out/r2c8_bench_cfg4.log 2>&1
tail -2 gpurun_out/r2c8_suite.log
tail -1 gpurun_out/r2c8_smoke.log
cat gpurun_out/r2c8_endurance.log
cat gpurun_out/r2c8_biggemm.log
grep -h '"value"' gpurun_out/r2c8_bench*.log
