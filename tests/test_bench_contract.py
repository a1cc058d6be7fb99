"""bench.py driver contract: JSON line format, flag handling, and the
exact multi-rank torchrun launch shape the round-end driver uses
(CPU/gloo here; RCCL on the GPU node)."""

import json
import os
import socket
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

TINY = ['--steps', '2', '--warmup', '1', '--batch', '2', '--classes', '4',
        '--mem', '8', '--mine', '4', '--img', '96', '--arch', 'resnet18',
        '--addon', 'regular']


def _parse_last_json(stdout):
    lines = [l for l in stdout.strip().splitlines() if l.startswith('{')]
    assert len(lines) == 1, stdout       # exactly ONE JSON line
    return json.loads(lines[-1])


def _check_contract(rec, n_gpus):
    assert rec['metric'] == 'train_images_per_sec'
    assert rec['unit'] == 'images/s'
    assert rec['n_gpus'] == n_gpus
    assert rec['steps'] == 2 and rec['warmup'] == 1
    assert rec['higher_is_better'] is True
    assert rec['scaling'] == 'weak'
    assert rec['vs_baseline'] is None
    assert rec['data'] == 'synthetic'
    assert rec['value'] > 0 and rec['ms_per_step'] > 0
    # whole-job value: images/s * ms_per_step ~= global batch images.
    # Both fields are rounded (2/3 decimals); at very low throughput (e.g.
    # CPU contention under pytest-xdist) the 0.005 img/s quantum dominates,
    # so scale the tolerance with it.
    per_step = rec['value'] * rec['ms_per_step'] / 1000.0
    tol = 0.01 + 0.005 / rec['value'] + 0.0005 / rec['ms_per_step']
    assert abs(per_step / rec['config']['global_batch'] - 1) < tol
    assert rec['config']['parallelism'] == f'dp{n_gpus}'


def test_bench_single_process():
    r = subprocess.run([sys.executable, 'bench.py', *TINY],
                       capture_output=True, text=True, timeout=600, cwd=ROOT)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    _check_contract(_parse_last_json(r.stdout), n_gpus=1)


def test_bench_torchrun_two_ranks():
    """The driver's exact launch shape at N=2 (gloo on CPU)."""
    with socket.socket() as s:
        s.bind(('127.0.0.1', 0))
        port = s.getsockname()[1]
    cmd = [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
           '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
           '--master-port', str(port), 'bench.py', '--gpus', '2', *TINY]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=900,
                       cwd=ROOT)
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    _check_contract(_parse_last_json(r.stdout), n_gpus=2)


def test_bench_preset_merge():
    """--preset applies the BASELINE config (arch/classes) while explicit
    flags still win."""
    cmd = [sys.executable, 'bench.py', '--preset', 'cars-densenet161',
           '--steps', '1', '--warmup', '0', '--batch', '2', '--mem', '8',
           '--mine', '4', '--img', '64']
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=900,
                       cwd=ROOT)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    rec = _parse_last_json(r.stdout)
    assert rec['config']['model'] == 'densenet161-mgproto'
    assert rec['config']['num_classes'] == 196      # preset
    assert rec['config']['global_batch'] == 2       # explicit flag wins


def test_bench_flags_reflected_in_config():
    """--no-em / --eager toggle the measured step and are reported."""
    cmd = [sys.executable, 'bench.py', *TINY, '--no-em', '--eager']
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                       cwd=ROOT)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    rec = _parse_last_json(r.stdout)
    assert rec['config']['em_active'] is False
    assert rec['config']['hip_graph'] is False
