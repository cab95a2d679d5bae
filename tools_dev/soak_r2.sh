#!/bin/bash
# Round-2 soak cycle: clean rebuild, full GPU suite, smoke, headline +
# epilogue bench, randomized parity sweep, CLI checks. Run repeatedly
# on fresh boxes; tag via $1.
set -x
R=$GRAFT_REPO_ROOT
T=${1:-soak}
mkdir -p $R/gpurun_out
cd $R
python -c 'import __graft_entry__; __graft_entry__.build()' > gpurun_out/${T}_build.log 2>&1
timeout 2100 python -m pytest tests -m gpu -q > gpurun_out/${T}_suite.log 2>&1
echo "suite rc=$?" >> gpurun_out/${T}_suite.log
python -c 'import __graft_entry__; __graft_entry__.smoke()' > gpurun_out/${T}_smoke.log 2>&1
timeout 400 python bench.py --steps 3 --warmup 1 --no-cpu-baseline > gpurun_out/${T}_bench.log 2>&1
timeout 400 python bench.py --workload epilogue --steps 2 --warmup 1 --no-cpu-baseline > gpurun_out/${T}_epi.log 2>&1
timeout 600 python tools_dev/parity_sweep.py 100 $RANDOM > gpurun_out/${T}_sweep.log 2>&1
./marlinx verify 300 200 100 > gpurun_out/${T}_cli.log 2>&1
./marlinx epilogue 200 160 180 >> gpurun_out/${T}_cli.log 2>&1
tail -2 gpurun_out/${T}_suite.log
tail -1 gpurun_out/${T}_smoke.log
grep -h '"value"' gpurun_out/${T}_bench.log gpurun_out/${T}_epi.log
tail -1 gpurun_out/${T}_sweep.log
cat gpurun_out/${T}_cli.log
