#!/usr/bin/env python3
"""Per-shape 1x1-conv A/B: MIOpen/CK conv vs hipBLASLt matmul (GPU).

Round-1 found routing ALL 1x1 convs through plain matmuls 35% slower
end-to-end; this measures per shape (fwd + input-grad + weight-grad,
bf16 NHWC, batch 80) whether a per-shape hybrid would beat either
global choice, and prints the winning table.

    gpurun -- 'python tools/bench_conv1x1.py'
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    from mgproto_amd.utils.helpers import setup_miopen_db
    setup_miopen_db()
    torch.backends.cudnn.benchmark = True
    dev = torch.device('cuda', 0)
    B = 80
    from tools.tune_miopen import collect_conv_shapes
    shapes, _ = collect_conv_shapes()
    one_by_one = [(cin, cout, st, H, W) for
                  (cin, cout, kh, kw, st, pad, H, W, g) in shapes
                  if kh == 1 and st == 1]
    print(f'{len(one_by_one)} stride-1 1x1 shapes')
    tot_conv, tot_mm, tot_hybrid = 0.0, 0.0, 0.0
    for cin, cout, st, H, W in one_by_one:
        x = torch.randn(B, cin, H, W, device=dev, dtype=torch.bfloat16) \
            .to(memory_format=torch.channels_last).requires_grad_(True)
        w = torch.randn(cout, cin, 1, 1, device=dev, dtype=torch.bfloat16) \
            .to(memory_format=torch.channels_last).requires_grad_(True)
        g = torch.randn(B, cout, H, W, device=dev, dtype=torch.bfloat16) \
            .to(memory_format=torch.channels_last)

        def conv_step():
            y = F.conv2d(x, w)
            y.backward(g)
            x.grad = None
            w.grad = None

        # matmul view: NHWC -> [M, Cin] @ [Cin, Cout]
        x2 = x.detach().permute(0, 2, 3, 1).reshape(-1, cin) \
            .contiguous().requires_grad_(True)
        w2 = w.detach().reshape(cout, cin).requires_grad_(True)
        g2 = g.permute(0, 2, 3, 1).reshape(-1, cout).contiguous()

        def mm_step():
            y = x2 @ w2.t()
            y.backward(g2)
            x2.grad = None
            w2.grad = None

        tc = timeit(conv_step)
        tm = timeit(mm_step)
        tot_conv += tc
        tot_mm += tm
        tot_hybrid += min(tc, tm)
        win = 'matmul' if tm < tc else 'conv'
        print(f'{cin:5d}x{H:3d}x{W:3d} -> {cout:5d}: conv {tc:7.3f} ms  '
              f'matmul {tm:7.3f} ms   -> {win}', flush=True)
    print(f'\ntotals (one call per unique shape): conv {tot_conv:.3f} ms, '
          f'matmul {tot_mm:.3f} ms, per-shape hybrid {tot_hybrid:.3f} ms')


if __name__ == '__main__':
    main()
