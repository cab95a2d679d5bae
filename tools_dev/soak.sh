#!/bin/bash
# One soak cycle (run ON the GPU box): full GPU suite + bench spot +
# one rotating extra check. Appends a one-line record to gpurun_out/.
set -e
R=${GRAFT_REPO_ROOT:-.}
cd $R
EXTRA=${1:-none}
SUITE=$(python -m pytest tests -m gpu -x -q 2>&1 | grep -E "passed|failed" | tail -1)
BENCH=$(timeout 150 python bench.py --steps 2 --warmup 1 --no-cpu-baseline 2>/dev/null | tail -1 | grep -o '"value": [0-9.]*' | head -1)
case $EXTRA in
  sweep)  X=$(timeout 600 python tools_dev/parity_sweep.py 60 2>&1 | tail -1);;
  f32)    X=$(MARLIN_BENCH_DTYPE=f32 MARLIN_BENCH_N=16384 timeout 100 python bench.py --steps 2 --warmup 1 --no-cpu-baseline 2>/dev/null | tail -1 | grep -o '"value": [0-9.]*');;
  smoke)  X=$(python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -1);;
  endure) X=$(timeout 300 python tools_dev/endurance.py 2>&1 | grep drift);;
  *)      X=none;;
esac
echo "soak $(date -u +%H:%M) rocm=$(ls /opt | grep -o 'rocm-[0-9.]*' | head -1) | $SUITE | $BENCH | extra[$EXTRA]: $X" | tee -a gpurun_out/soak.log
