#!/bin/bash
# Round-2 call 4: ASAN run record, k-phase mask-8 paired experiment,
# full GPU suite, and the 2-ranks-on-1-GPU RCCL probe (bounded; RCCL
# may refuse multiple ranks per device — an error is itself the answer).
set -x
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out
cd $R

# --- ASAN build + targeted GPU tests (SURVEY §5 sanitizer plan) -------
ASAN_RT=$(find /opt/rocm/lib/llvm -name "libclang_rt.asan-x86_64.so" | head -1)
MARLIN_ASAN=1 python -c 'import __graft_entry__; __graft_entry__.build()' \
  > gpurun_out/r2c4_asan_build.log 2>&1
LD_PRELOAD=$ASAN_RT ASAN_OPTIONS=detect_leaks=0 \
  timeout 600 python -m pytest tests/test_gpu_parity.py -q -m gpu \
  -k "golden or tile or summa_single or kres or error_paths or rccl or zero_pad" \
  > gpurun_out/r2c4_asan_tests.log 2>&1
echo "asan pytest rc=$?" >> gpurun_out/r2c4_asan_tests.log

# --- normal rebuild + FULL GPU suite ----------------------------------
python -c 'import __graft_entry__; __graft_entry__.build()' \
  > gpurun_out/r2c4_build.log 2>&1
timeout 2100 python -m pytest tests -m gpu -q \
  > gpurun_out/r2c4_gputests.log 2>&1
echo "pytest rc=$?" >> gpurun_out/r2c4_gputests.log

# --- paired phase experiment (mask 8 = true per-CU stagger) -----------
B="python bench.py --gpus 1 --warmup 1 --no-cpu-baseline"
timeout 400 $B --steps 3 > gpurun_out/r2c4_f64_base.log 2>&1
MARLIN_GEMM_PHASE=8 timeout 400 $B --steps 3 > gpurun_out/r2c4_f64_ph8.log 2>&1
MARLIN_GEMM_PHASE=1 timeout 400 $B --steps 3 > gpurun_out/r2c4_f64_ph1.log 2>&1

# --- aux kernel bandwidth after gemv 32-chunk fix ---------------------
timeout 300 python tools_dev/bench_aux_device.py \
  > gpurun_out/r2c4_aux.log 2>&1

# --- 2 ranks on 1 GPU (RCCL multi-rank smoke; bounded) ----------------
export HSA_ENABLE_IPC_MODE_LEGACY=0
MARLIN_FORCE_DEV0=1 MARLIN_BENCH_N=4096 MARLIN_SUMMA_DEBUG=1 \
  timeout 240 python -m torch.distributed.run --nnodes=1 \
  --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29431 \
  bench.py --gpus 2 --steps 2 --warmup 1 --no-cpu-baseline \
  > gpurun_out/r2c4_2rank.log 2>&1
echo "2rank rc=$?" >> gpurun_out/r2c4_2rank.log

tail -3 gpurun_out/r2c4_asan_tests.log
tail -3 gpurun_out/r2c4_gputests.log
grep -h '"value"' gpurun_out/r2c4_f64_*.log
tail -6 gpurun_out/r2c4_2rank.log
