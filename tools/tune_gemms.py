#!/usr/bin/env python3
"""Generate (and ship) TunableOp-tuned GEMM configs for the flagship step.

hipBLASLt's default heuristics lose to MIOpen's CK picks on this model's
skinny-K 1x1-conv GEMMs (profiles/README.md). PyTorch TunableOp benchmarks
every GEMM shape it sees and records the best solution in a CSV; run this
ON A GPU BOX, then commit the CSV under mgproto_amd/tunableop/ — bench.py
and train.py point PYTORCH_TUNABLEOP_FILENAME at a writable copy the same
way the MIOpen find-db is shipped.

    gpurun -- 'python tools/tune_gemms.py --steps 3'
    cp gpurun_out/tunableop_results*.csv mgproto_amd/tunableop/
"""

import argparse
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--steps', type=int, default=3)
    ap.add_argument('--out', type=str, default='gpurun_out')
    args = ap.parse_args()

    os.makedirs(args.out, exist_ok=True)
    env = dict(os.environ)
    env['PYTORCH_TUNABLEOP_ENABLED'] = '1'
    env['PYTORCH_TUNABLEOP_TUNING'] = '1'
    env['PYTORCH_TUNABLEOP_FILENAME'] = os.path.join(
        args.out, 'tunableop_results.csv')
    # tune with the GEMM conv path active so the 1x1-conv shapes are seen
    env['MGPROTO_GEMM_CONV1X1'] = '1'
    cmd = [sys.executable, os.path.join(ROOT, 'bench.py'),
           '--steps', str(args.steps), '--warmup', '2', '--no-graph']
    print('tuning with:', ' '.join(cmd))
    r = subprocess.run(cmd, env=env, cwd=ROOT)
    print('tunableop CSV at', env['PYTORCH_TUNABLEOP_FILENAME'])
    sys.exit(r.returncode)


if __name__ == '__main__':
    main()
