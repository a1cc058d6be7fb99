#!/bin/bash
# Synchronization-bug shakeout (run on an MI355X box):
# serializes every kernel launch and copy so missing stream/event
# dependencies surface as wrong results instead of latent races, then
# re-runs the GPU suite and the determinism tests twice.
#
#   /usr/local/graft/bin/gpurun --timeout 1800 -- 'bash scripts/sanitize.sh'
set -x
export AMD_SERIALIZE_KERNEL=3      # launch-serialize + sync after each kernel
export AMD_SERIALIZE_COPY=3
export HIP_LAUNCH_BLOCKING=1
mkdir -p gpurun_out
timeout 1200 python -m pytest tests -m gpu -q > gpurun_out/sanitize_pass1.log 2>&1
echo "pass1: $?"
timeout 600 python -m pytest tests/test_determinism_gpu.py -q > gpurun_out/sanitize_pass2.log 2>&1
echo "pass2: $?"
