#!/usr/bin/env python3
"""Per-shape A/B: the hand-written bf16 MFMA GEMM (gemm1x1_fwd) vs
MIOpen/CK F.conv2d forward on the flagship 1x1 shapes (GPU).

    gpurun -- 'MGPROTO_GEMM1X1_HIP=1 python tools/bench_gemm1x1_hip.py'
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    from mgproto_amd.ops import hip_loader
    from mgproto_amd.utils.helpers import setup_miopen_db
    setup_miopen_db()
    torch.backends.cudnn.benchmark = True
    ext = hip_loader.load()
    dev = torch.device('cuda', 0)
    B = 80
    shapes = [(64, 256, 112), (256, 64, 112), (64, 64, 112),
              (128, 512, 56), (512, 128, 56),
              (256, 1024, 28), (1024, 256, 28),
              (512, 2048, 14), (2048, 512, 14), (2048, 64, 14)]
    tot_ck, tot_mine, tot_mine_bn = 0.0, 0.0, 0.0
    for K, N, H in shapes:
        M = B * H * H
        x = torch.randn(M, K, device=dev).bfloat16() / (K ** 0.5)
        w = torch.randn(N, K, device=dev).bfloat16() / (K ** 0.5)
        x4 = x.view(B, H, H, K).permute(0, 3, 1, 2) \
            .contiguous(memory_format=torch.channels_last)
        w4 = w.view(N, K, 1, 1).contiguous(memory_format=torch.channels_last)

        t_ck = timeit(lambda: F.conv2d(x4, w4))
        t_me = timeit(lambda: ext.gemm1x1_fwd(x, w, None, False))
        t_bn = timeit(lambda: ext.gemm1x1_fwd(x, w, None, True))
        t_md = {m: timeit(lambda m=m: ext.gemm1x1_fwd(x, w, None, False, m))
                for m in (0, 1, 2)}
        # correctness spot-checks on the fly (auto + the glds mode)
        want = (x.float() @ w.float().t()).bfloat16().float()
        y, _ = ext.gemm1x1_fwd(x, w, None, False)
        ok = torch.allclose(y.float(), want, rtol=2e-2, atol=1e-2)
        y2, _ = ext.gemm1x1_fwd(x, w, None, False, 2)
        ok2 = torch.allclose(y2.float(), want, rtol=2e-2, atol=1e-2)
        tot_ck += t_ck; tot_mine += t_me; tot_mine_bn += t_bn
        gb = (M * K + K * N + M * N) * 2 / 1e9
        print(f'K={K:5d} N={N:5d} M={M:7d}: CK {t_ck:7.3f} ms | '
              f'auto {t_me:7.3f} ms ({gb/t_me*1e3:5.0f} GB/s) | '
              f'plain {t_md[0]:7.3f} dbuf {t_md[1]:7.3f} '
              f'glds {t_md[2]:7.3f} | +BNstats {t_bn:7.3f} | '
              f'parity={"OK" if ok else "FAIL"}/{"OK" if ok2 else "FAIL"}',
              flush=True)
    print(f'\ntotals: CK {tot_ck:.3f} ms, mine {tot_mine:.3f} ms, '
          f'mine+BNstats {tot_mine_bn:.3f} ms')


if __name__ == '__main__':
    main()
