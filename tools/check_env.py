#!/usr/bin/env python3
"""Environment diagnostic: verifies the stack this framework needs
(ROCm torch, gfx950 extension, RCCL backend, MIOpen find-db, native
augmentation core) and prints one PASS/FAIL line per item."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    checks = []

    import torch
    checks.append(('torch', True, torch.__version__))
    is_rocm = torch.version.hip is not None
    checks.append(('ROCm build', is_rocm, torch.version.hip or 'CUDA/cpu build'))
    has_gpu = torch.cuda.is_available()
    name = torch.cuda.get_device_name(0) if has_gpu else 'none'
    checks.append(('GPU', has_gpu, name))
    if has_gpu:
        arch = torch.cuda.get_device_properties(0).gcnArchName
        checks.append(('gfx950 (MI355X)', 'gfx950' in arch, arch))

    from mgproto_amd.ops import native_available
    from mgproto_amd.ops.hip_loader import SO_PATH, CPU_SO_PATH
    built = os.path.isfile(SO_PATH)
    checks.append(('HIP extension built', built, SO_PATH))
    if has_gpu:
        checks.append(('HIP extension loads', native_available(), ''))
    checks.append(('CPU augmentation core', os.path.isfile(CPU_SO_PATH),
                   CPU_SO_PATH))

    import torch.distributed as dist
    checks.append(('RCCL (nccl) backend', dist.is_nccl_available(), ''))
    checks.append(('gloo backend', dist.is_gloo_available(), ''))

    db = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), 'mgproto_amd', 'miopen_db')
    have_db = os.path.isdir(db) and bool(os.listdir(db))
    checks.append(('MIOpen find-db shipped', have_db, db))
    checks.append(('HSA_ENABLE_IPC_MODE_LEGACY=0 (multi-proc IPC)',
                   os.environ.get('HSA_ENABLE_IPC_MODE_LEGACY') == '0',
                   os.environ.get('HSA_ENABLE_IPC_MODE_LEGACY', 'unset')))

    width = max(len(n) for n, _, _ in checks)
    fails = 0
    for n, ok, detail in checks:
        fails += not ok
        print(f'{"PASS" if ok else "FAIL"}  {n:<{width}}  {detail}')
    print(f'{len(checks) - fails}/{len(checks)} checks passed')
    return 0 if fails == 0 else 1


if __name__ == '__main__':
    raise SystemExit(main())
