#!/bin/bash
# Round-2 GPU call 2: graph-trainer validation, enqueue E2E debug,
# suite under the new defaults (BN mask on), EM-alone bench A/B.
#   /usr/local/graft/bin/gpurun --timeout 1500 -- 'bash tools/round2_call2.sh'
set -x
mkdir -p gpurun_out/r2b

# 1. new hipGraph-trainer tests
timeout 600 python -m pytest tests/test_graph_train_gpu.py -x -q \
    > gpurun_out/r2b/graph_tests.log 2>&1
echo "graph: $?" >> gpurun_out/r2b/summary.txt

# 2. localize the HIP-enqueue E2E divergence
timeout 300 python tools/debug_enqueue_e2e.py \
    > gpurun_out/r2b/enqueue_debug.log 2>&1
echo "enqdbg: $?" >> gpurun_out/r2b/summary.txt

# 3. full GPU suite under the new defaults
timeout 900 python -m pytest tests -m gpu -q \
    > gpurun_out/r2b/pytest_gpu.log 2>&1
echo "pytest: $?" >> gpurun_out/r2b/summary.txt

# 4. benches on THIS box: default (graph, BN-mask on) and EM-HIP alone
timeout 600 python bench.py --steps 100 --warmup 10 \
    > gpurun_out/r2b/bench_default.json 2>gpurun_out/r2b/bench_default.log
echo "bench: $?" >> gpurun_out/r2b/summary.txt
timeout 600 bash -c 'MGPROTO_HIP_EM=1 python bench.py --steps 100 --warmup 10' \
    > gpurun_out/r2b/bench_em.json 2>/dev/null
echo "bench-em: $?" >> gpurun_out/r2b/summary.txt
echo done >> gpurun_out/r2b/summary.txt
