#!/bin/bash
# Round-1 validation bundle (run ON the GPU box).
set -x
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out
cd $R

# 1) torchrun single-rank (the exact driver launch shape for N>1)
timeout 240 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 \
  --master-addr 127.0.0.1 --master-port 29511 \
  bench.py --gpus 1 --steps 2 --warmup 1 --n 4096 --no-cpu-baseline \
  > gpurun_out/torchrun1.log 2>&1
echo "TR1=$?" >> gpurun_out/torchrun1.log

# 2) two ranks on one GPU (RCCL may refuse duplicate device: informative)
timeout 240 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29512 \
  bench.py --gpus 2 --steps 1 --warmup 0 --n 2048 --no-cpu-baseline \
  > gpurun_out/torchrun2.log 2>&1
echo "TR2=$?" >> gpurun_out/torchrun2.log

# 3) config 4 tall-skinny and config 5 fp32
timeout 300 python bench.py --gpus 1 --steps 2 --warmup 1 \
  --m 50000 --k 4096 --nn 50000 --no-cpu-baseline \
  > gpurun_out/bench_cfg4.log 2>&1
timeout 300 python bench.py --gpus 1 --steps 2 --warmup 1 \
  --n 40000 --dtype f32 --no-cpu-baseline \
  > gpurun_out/bench_cfg5.log 2>&1

# 4) PMC: FETCH_SIZE + GRBM (effective clock) at 20000^3, separate passes
cd /tmp && export TMPDIR=/tmp
B="python $R/bench.py --gpus 1 --steps 1 --warmup 1 --n 20000 --no-cpu-baseline"
rocprofv3 --pmc FETCH_SIZE -d $R/gpurun_out/pmc_fetch2 -o fetch2 \
  --output-format csv -- $B > $R/gpurun_out/pmc_fetch2.log 2>&1
rocprofv3 --pmc GRBM_GUI_ACTIVE GRBM_COUNT -d $R/gpurun_out/pmc_grbm \
  -o grbm --output-format csv -- $B > $R/gpurun_out/pmc_grbm.log 2>&1

# 5) kernel-trace stats refresh (rocpd db)
rocprofv3 --kernel-trace --stats -d $R/gpurun_out/prof20000b -- \
  python $R/bench.py --gpus 1 --steps 2 --warmup 1 --n 20000 \
  --no-cpu-baseline > $R/gpurun_out/prof20000b.log 2>&1

tail -1 $R/gpurun_out/torchrun1.log
tail -3 $R/gpurun_out/torchrun2.log
tail -1 $R/gpurun_out/bench_cfg4.log
tail -1 $R/gpurun_out/bench_cfg5.log
