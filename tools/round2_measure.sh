#!/bin/bash
# One-call GPU measurement bundle (run via gpurun on an MI355X box):
#   /usr/local/graft/bin/gpurun --timeout 2700 -- 'bash tools/round2_measure.sh'
# Writes everything under gpurun_out/r2/ for merge-back.
set -x
mkdir -p gpurun_out/r2

# 1. full GPU test suite
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/r2/pytest_gpu.log 2>&1
echo "pytest: $?" >> gpurun_out/r2/summary.txt

# 2. BN-mask experimental path: parity test + bench A/B
timeout 600 bash -c 'MGPROTO_BN_MASK=1 python -m pytest tests/test_fused_bn_gpu.py -x -q' \
    > gpurun_out/r2/bn_mask_tests.log 2>&1
echo "bn-mask tests: $?" >> gpurun_out/r2/summary.txt

# 2b. HIP EM kernels: parity tests + A/B bench
timeout 600 bash -c 'MGPROTO_HIP_EM=1 python -m pytest tests/test_em_hip_gpu.py -x -q' \
    > gpurun_out/r2/em_hip_tests.log 2>&1
echo "em-hip tests: $?" >> gpurun_out/r2/summary.txt
timeout 600 bash -c 'MGPROTO_HIP_ENQUEUE=1 python -m pytest tests/test_enqueue_hip_gpu.py -x -q' \
    > gpurun_out/r2/enqueue_hip_tests.log 2>&1
echo "enqueue-hip tests: $?" >> gpurun_out/r2/summary.txt
timeout 600 bash -c 'MGPROTO_HIP_EM=1 MGPROTO_HIP_ENQUEUE=1 python bench.py --steps 30 --warmup 10' > gpurun_out/r2/bench_emhip.json 2>/dev/null

# 3. flagship bench: default, BN-mask on, eager (for the graph delta)
timeout 600 python bench.py --steps 30 --warmup 10 > gpurun_out/r2/bench_default.json 2>gpurun_out/r2/bench_default.log
timeout 600 bash -c 'MGPROTO_BN_MASK=1 python bench.py --steps 30 --warmup 10' > gpurun_out/r2/bench_bnmask.json 2>/dev/null
timeout 600 python bench.py --steps 30 --warmup 10 --no-graph > gpurun_out/r2/bench_nograph.json 2>/dev/null

# 3b. serving latency (eager vs captured forward)
timeout 600 python tools/serve_bench.py --batch 8 --iters 50 > gpurun_out/r2/serve_bench.jsonl 2>/dev/null

# 4. kernel-level profile of the default config (small CSVs only)
cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
timeout 900 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof_r2 \
    -- python bench.py --steps 10 --warmup 8 > gpurun_out/r2/bench_profiled.json 2>gpurun_out/r2/prof.log
find /tmp/prof_r2 -name '*stats*.csv' -size -2M -exec cp {} gpurun_out/r2/ \; 2>/dev/null
echo "done" >> gpurun_out/r2/summary.txt
