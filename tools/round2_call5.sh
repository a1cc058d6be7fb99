#!/bin/bash
# Round-2 GPU call 5: final validation — upsample kernels + contiguity
# fix + full suite, final bench, per-shape 1x1 A/B, DenseNet-161 preset
# (longer find window), serialized-kernel sanitize pass, rocprof stats.
#   /usr/local/graft/bin/gpurun --timeout 2000 -- 'bash tools/round2_call5.sh'
set -x
mkdir -p gpurun_out/r2e

timeout 700 python -m pytest tests -m gpu -q > gpurun_out/r2e/pytest_gpu.log 2>&1
echo "pytest: $?" >> gpurun_out/r2e/summary.txt

timeout 400 python bench.py --steps 100 --warmup 10 \
    > gpurun_out/r2e/bench_final.json 2>/dev/null
echo "bench: $?" >> gpurun_out/r2e/summary.txt

timeout 400 python tools/bench_conv1x1.py > gpurun_out/r2e/conv1x1.log 2>&1
echo "conv1x1: $?" >> gpurun_out/r2e/summary.txt

timeout 900 python bench.py --preset cars-densenet161 --steps 30 --warmup 10 \
    > gpurun_out/r2e/bench_dn161.json 2>gpurun_out/r2e/bench_dn161.log
echo "dn161: $?" >> gpurun_out/r2e/summary.txt

# serialized-kernel shakeout: missing stream/event deps
# surface as wrong results when every launch synchronizes
AMD_SERIALIZE_KERNEL=3 AMD_SERIALIZE_COPY=3 HIP_LAUNCH_BLOCKING=1 \
timeout 600 python -m pytest tests/test_ops_gpu.py tests/test_fused_bn_gpu.py \
    tests/test_determinism_gpu.py -q > gpurun_out/r2e/sanitize.log 2>&1
echo "sanitize: $?" >> gpurun_out/r2e/summary.txt

cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof_r2e \
    -- python bench.py --steps 10 --warmup 8 > gpurun_out/r2e/bench_profiled.json \
    2>gpurun_out/r2e/prof.log
find /tmp/prof_r2e -name '*stats*.csv' -size -2M -exec cp {} gpurun_out/r2e/ \; 2>/dev/null
echo "prof: $?" >> gpurun_out/r2e/summary.txt
echo done >> gpurun_out/r2e/summary.txt
