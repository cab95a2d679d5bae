#!/bin/bash
# Round-2 call 3: HBM traffic evidence — rocprofv3
# FETCH_SIZE and WRITE_SIZE passes (separate passes: TCC slots cannot
# fit both) for configs 2/4/5, the epilogue leg, and the aux HBM-bound
# kernels (+ fill_random as the WRITE_SIZE calibration: it writes a
# known byte count and reads ~nothing). Plus a kernel-trace pass for
# the epilogue leg's committed kernel stats.
set -x
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out/r2pmc
cd /tmp && export TMPDIR=/tmp

python -c 'import sys; sys.path.insert(0,"'$R'"); import __graft_entry__; __graft_entry__.build()' \
  > $R/gpurun_out/r2pmc/build.log 2>&1

B="python $R/bench.py --gpus 1 --steps 1 --warmup 1 --no-cpu-baseline"
pass() { # pass TAG COUNTER -- cmd...
  local TAG=$1 CTR=$2; shift 2
  rocprofv3 --pmc $CTR -d $R/gpurun_out/r2pmc/$TAG -o $TAG \
    --output-format csv -- "$@" > $R/gpurun_out/r2pmc/$TAG.log 2>&1
}

# config 2: 10000^3 fp64
pass cfg2_f  FETCH_SIZE $B --n 10000
pass cfg2_w  WRITE_SIZE $B --n 10000
# config 3 size (20000^3): WRITE pass (FETCH committed in r01)
pass cfg3_w  WRITE_SIZE $B
# config 4: 50000x4096x50000 fp64
pass cfg4_f  FETCH_SIZE $B --m 50000 --k 4096 --nn 50000
pass cfg4_w  WRITE_SIZE $B --m 50000 --k 4096 --nn 50000
# config 5 plain fp32 40000^2
pass cfg5_f  FETCH_SIZE $B --dtype f32 --n 40000
pass cfg5_w  WRITE_SIZE $B --dtype f32 --n 40000
# epilogue leg
pass epi_f   FETCH_SIZE $B --workload epilogue
pass epi_w   WRITE_SIZE $B --workload epilogue
# aux kernels (+ fill_random write calibration inside)
AUX="python $R/tools_dev/bench_aux_device.py"
pass aux_f FETCH_SIZE $AUX
pass aux_w WRITE_SIZE $AUX

# epilogue kernel-trace stats
rocprofv3 --kernel-trace --stats -d $R/gpurun_out/r2pmc/epi_trace -- \
  $B --workload epilogue --steps 2 > $R/gpurun_out/r2pmc/epi_trace.log 2>&1

# summarise every CSV per kernel
cd $R
for f in $(find gpurun_out/r2pmc -name "*counter_collection.csv"); do
  echo "== $f"
  python tools_dev/rocpd_stats.py traffic $f
done > gpurun_out/r2pmc/summary.txt 2>&1
for db in $(find gpurun_out/r2pmc/epi_trace -name "*.db"); do
  python tools_dev/rocpd_stats.py kernel $db
done >> gpurun_out/r2pmc/summary.txt 2>&1
tail -40 gpurun_out/r2pmc/summary.txt
