#!/usr/bin/env python3
"""Find which part of the training step blocks hipGraph capture."""

import os
import sys
import traceback

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from mgproto_amd.model import construct_MGProto
from mgproto_amd.losses import build_aux_loss


def try_capture(name, fn, warm=3):
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    try:
        with torch.cuda.stream(s):
            for _ in range(warm):
                fn()
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            fn()
        g.replay()
        torch.cuda.synchronize()
        print(f'[OK]   {name}')
        return True
    except Exception:  # noqa: BLE001  (any capture failure is the signal)
        torch.cuda.synchronize()
        print(f'[FAIL] {name}')
        traceback.print_exc(limit=8)
        return False


def main():
    if not torch.cuda.is_available():
        print('graphprobe: GPU-only diagnostic (hipGraph capture probe); '
              'no HIP device found.')
        return
    from mgproto_amd.utils.helpers import setup_miopen_db
    setup_miopen_db()
    torch.backends.cudnn.benchmark = True
    dev = torch.device('cuda', 0)
    C, K, d = 200, 10, 64
    torch.manual_seed(0)
    model = construct_MGProto('resnet50', pretrained=False, img_size=224,
                              prototype_shape=(C * K, d, 1, 1), num_classes=C,
                              add_on_layers_type='regular_upsample',
                              sz_embedding=32, mem_capacity=800,
                              mine_K=20).to(dev)
    model.features = model.features.to(memory_format=torch.channels_last)
    aux = build_aux_loss('Proxy_Anchor', nb_classes=C, sz_embed=32).to(dev)
    try:
        opt = torch.optim.Adam(model.parameters(), lr=1e-4, fused=True,
                               capturable=True)
    except Exception as e:  # noqa: BLE001
        print('fused adam failed:', e)
        opt = torch.optim.Adam(model.parameters(), lr=1e-4, capturable=True)
    model.train()

    x = torch.randn(80, 3, 224, 224, device=dev).contiguous(
        memory_format=torch.channels_last)
    gt = torch.randint(0, C, (80,), device=dev)
    with torch.no_grad():
        mem = F.normalize(torch.randn(C * 800, d, device=dev), dim=1)
        labels = torch.arange(C, device=dev).repeat_interleave(800)
        model.queue.push(mem, labels)

    amp = lambda: torch.autocast('cuda', dtype=torch.bfloat16)  # noqa: E731

    def bb():
        with torch.no_grad(), amp():
            model.features(x)
    try_capture('backbone fwd', bb)

    def full_fwd_nogt():
        with torch.no_grad(), amp():
            model(x, None)
    try_capture('model fwd (no gt)', full_fwd_nogt)

    def full_fwd():
        with torch.no_grad(), amp():
            model(x, gt)
    try_capture('model fwd (gt, enqueue)', full_fwd)

    def fwd_bwd():
        with amp():
            out, emb = model(x, gt)
        loss = F.cross_entropy(out.float()[:, :, 0], gt)
        model.zero_grad(set_to_none=True)
        loss.backward()
    try_capture('fwd+bwd', fwd_bwd)

    def fwd_bwd_aux():
        with amp():
            out, emb = model(x, gt)
        out = out.float()
        mine = sum(F.cross_entropy(out[:, :, k], gt)
                   for k in range(1, out.shape[2])) / (out.shape[2] - 1)
        loss = F.cross_entropy(out[:, :, 0], gt) + 0.2 * mine \
            + 0.5 * aux(emb.float(), gt)
        model.zero_grad(set_to_none=True)
        loss.backward()
    try_capture('fwd+bwd+losses', fwd_bwd_aux)

    def with_opt():
        fwd_bwd_aux()
        opt.step()
    try_capture('fwd+bwd+opt', with_opt)

    def with_em():
        model.memory_updated_cls[:] = True
        model.update_GMM()
    try_capture('EM', with_em)

    def everything():
        with_opt()
        with_em()
    try_capture('full step', everything)


if __name__ == '__main__':
    main()
