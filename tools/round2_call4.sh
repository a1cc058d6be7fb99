#!/bin/bash
# Round-2 GPU call 4: stage-2 BN A/B, uni-GMM validation, TunableOp GEMM
# tuning + 1x1 re-evaluation, BASELINE preset benches.
#   /usr/local/graft/bin/gpurun --timeout 2100 -- 'bash tools/round2_call4.sh'
set -x
mkdir -p gpurun_out/r2d

# 1. GPU suite (new uni-GMM parity + eval drivers + BN changes)
timeout 900 python -m pytest tests -m gpu -q \
    > gpurun_out/r2d/pytest_gpu.log 2>&1
echo "pytest: $?" >> gpurun_out/r2d/summary.txt

# 2. BN per-shape A/B + whole-step bench
timeout 300 python tools/bench_bn.py > gpurun_out/r2d/bench_bn.log 2>&1
timeout 600 python bench.py --steps 100 --warmup 10 \
    > gpurun_out/r2d/bench_default.json 2>/dev/null
echo "bench: $?" >> gpurun_out/r2d/summary.txt

# 3. TunableOp GEMM tuning for the 1x1-conv shapes, then A/B
timeout 900 python tools/tune_gemms.py --steps 2 --out gpurun_out/r2d \
    > gpurun_out/r2d/tune_gemms.log 2>&1
echo "tunegemm: $?" >> gpurun_out/r2d/summary.txt
timeout 600 bash -c 'MGPROTO_GEMM_CONV1X1=1 PYTORCH_TUNABLEOP_ENABLED=1 \
    PYTORCH_TUNABLEOP_TUNING=0 \
    PYTORCH_TUNABLEOP_FILENAME=gpurun_out/r2d/tunableop_results.csv \
    python bench.py --steps 50 --warmup 10' \
    > gpurun_out/r2d/bench_gemm1x1_tuned.json 2>/dev/null
echo "bench-gemm: $?" >> gpurun_out/r2d/summary.txt

# 4. BASELINE preset benches (configs 3-5)
for p in cars-densenet161 ood-resnet152 pets-vgg19; do
    timeout 600 python bench.py --preset $p --steps 50 --warmup 10 \
        > gpurun_out/r2d/bench_$p.json 2>gpurun_out/r2d/bench_$p.log
    echo "preset-$p: $?" >> gpurun_out/r2d/summary.txt
done
echo done >> gpurun_out/r2d/summary.txt
