#!/bin/bash
# Round-2 GPU call 3: BN-unroll A/B + fixed tests + MIOpen SEARCH tuning.
#   /usr/local/graft/bin/gpurun --timeout 2400 -- 'bash tools/round2_call3.sh'
set -x
mkdir -p gpurun_out/r2c

# 1. fused-BN parity (the unrolled reduction kernels must stay exact)
timeout 600 python -m pytest tests/test_fused_bn_gpu.py tests/test_enqueue_hip_gpu.py tests/test_graph_train_gpu.py -q \
    > gpurun_out/r2c/tests.log 2>&1
echo "tests: $?" >> gpurun_out/r2c/summary.txt

# 2. per-shape BN bandwidth (A/B against round-2 call-1 numbers)
timeout 300 python tools/bench_bn.py > gpurun_out/r2c/bench_bn.log 2>&1
echo "bn: $?" >> gpurun_out/r2c/summary.txt

# 3. whole-step bench with the unrolled BN
timeout 600 python bench.py --steps 100 --warmup 10 \
    > gpurun_out/r2c/bench_unroll.json 2>/dev/null
echo "bench: $?" >> gpurun_out/r2c/summary.txt

# 4. MIOpen SEARCH auto-tune (incremental db; budget-bounded), then re-bench
timeout 1400 python tools/tune_miopen.py --budget 1100 \
    --db gpurun_out/r2c/miopen_tuned --seed-from-repo \
    > gpurun_out/r2c/tune.log 2>&1
echo "tune: $?" >> gpurun_out/r2c/summary.txt
timeout 600 bash -c 'MIOPEN_USER_DB_PATH=$PWD/gpurun_out/r2c/miopen_tuned python bench.py --steps 100 --warmup 10' \
    > gpurun_out/r2c/bench_tuned.json 2>/dev/null
echo "bench-tuned: $?" >> gpurun_out/r2c/summary.txt
echo done >> gpurun_out/r2c/summary.txt
