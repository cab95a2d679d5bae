#!/bin/bash
# PMC counter passes for the fp64 GEMM kernel (run ON the GPU box).
# Collects SQ wave/wait/LDS counters and TCC FETCH_SIZE in separate
# passes (gfx950 slot limits; never combined with trace domains).
set -x
cd /tmp && export TMPDIR=/tmp
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out

rocprofv3 -L 2>/dev/null | grep -oE "SQ_[A-Z_0-9]+|TCC_[A-Z_0-9]+" | sort -u \
  > $R/gpurun_out/pmc_names.txt
head -100 $R/gpurun_out/pmc_names.txt

BENCH="python $R/bench.py --gpus 1 --steps 2 --warmup 1 --n 8192 --no-cpu-baseline"

# pass 1: SQ issue/wait/LDS
rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
    SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE SQ_ACTIVE_INST_ANY \
    -d $R/gpurun_out/pmc_sq -o sq --output-format csv -- $BENCH \
    > $R/gpurun_out/pmc_sq.log 2>&1
# pass 2: MFMA busy + waves
rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVES SQ_WAIT_INST_LDS \
    -d $R/gpurun_out/pmc_mfma -o mfma --output-format csv -- $BENCH \
    > $R/gpurun_out/pmc_mfma.log 2>&1
# pass 3: HBM fetch
rocprofv3 --pmc FETCH_SIZE \
    -d $R/gpurun_out/pmc_fetch -o fetch --output-format csv -- $BENCH \
    > $R/gpurun_out/pmc_fetch.log 2>&1
find $R/gpurun_out/pmc_sq $R/gpurun_out/pmc_mfma $R/gpurun_out/pmc_fetch -name "*.csv" | head
