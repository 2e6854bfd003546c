#!/bin/bash
# GPU validation + GEMM tuning experiment
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

# 1. GPU test suite
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/gputests.log 2>&1
echo "pytest rc=$?" >> gpurun_out/gputests.log

# 2. baseline bench (B=8) for reference on this box
timeout 420 python bench.py --steps 6 --warmup 2 > gpurun_out/bench_b8.json 2> gpurun_out/bench_b8.err

# 3. global batch 16
timeout 420 python bench.py --steps 6 --warmup 2 --global-batch 16 > gpurun_out/bench_b16.json 2> gpurun_out/bench_b16.err

# 4. TunableOp: tune GEMM algos (no capture, short), then timed run with tuned table
export PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tunableop.csv
PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 HETU_AMD_CAPTURE=0 \
  timeout 1200 python bench.py --steps 2 --warmup 1 > gpurun_out/bench_tuning.json 2> gpurun_out/bench_tuning.err
ls gpurun_out/tunableop* >> gpurun_out/gputests.log 2>&1
PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=0 \
  timeout 420 python bench.py --steps 6 --warmup 2 > gpurun_out/bench_tuned_b8.json 2> gpurun_out/bench_tuned_b8.err
PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=0 \
  timeout 420 python bench.py --steps 6 --warmup 2 --global-batch 16 > gpurun_out/bench_tuned_b16.json 2> gpurun_out/bench_tuned_b16.err
echo ALLDONE
