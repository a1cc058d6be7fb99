#!/usr/bin/env python3
"""Learning-equivalence A/B: reference MGProto vs mgproto_amd.

Runs the ORIGINAL reference implementation (imported from
/root/reference, pure PyTorch) next to this framework on identical
synthetic data, identical initial weights (loaded through the
checkpoint-interop path), the same optimizer spec and schedule — and
records the per-step loss/accuracy trajectories side by side
(VERDICT.md round-1 missing #2: "same model, faster" must be shown, not
asserted).

The reference is treated as an oracle binary: we stub its unavailable
imports (cv2, torchvision, pytorch_metric_learning — none touch the
math on this path) and no-op ``Tensor.cuda`` on CPU-only hosts (the
reference hardcodes ``.cuda()`` in `_m_step_diversified` model.py:391
and Proxy_Anchor utils/losses.py:33). No reference code is copied.

Usage:
    python tools/ab_reference.py --steps 32 --out profiles/ab_reference.md
"""

import argparse
import importlib
import os
import sys
import types

import torch
import torch.nn.functional as F

_REF = None


def load_reference(path='/root/reference'):
    """Import the reference's model + losses modules with import stubs."""
    global _REF
    if _REF is not None:
        return _REF
    # stubs for modules absent in this image; the A/B path never calls them
    if 'cv2' not in sys.modules:
        sys.modules['cv2'] = types.ModuleType('cv2')
    if 'torchvision' not in sys.modules:
        tv = types.ModuleType('torchvision')
        tvd = types.ModuleType('torchvision.datasets')

        class _ImageFolder:  # placeholder base class for MyImageFolder
            pass

        tvd.ImageFolder = _ImageFolder
        tv.datasets = tvd
        sys.modules['torchvision'] = tv
        sys.modules['torchvision.datasets'] = tvd
    if 'pytorch_metric_learning' not in sys.modules:
        pml = types.ModuleType('pytorch_metric_learning')
        pml.miners = types.ModuleType('pytorch_metric_learning.miners')
        pml.losses = types.ModuleType('pytorch_metric_learning.losses')
        sys.modules['pytorch_metric_learning'] = pml
    if 'matplotlib' not in sys.modules:
        mpl = types.ModuleType('matplotlib')
        mpl.pyplot = types.ModuleType('matplotlib.pyplot')
        sys.modules['matplotlib'] = mpl
        sys.modules['matplotlib.pyplot'] = mpl.pyplot
    if not torch.cuda.is_available():
        # reference hardcodes .cuda() on a few tensors; make it identity
        torch.Tensor.cuda = lambda self, *a, **k: self  # type: ignore[assignment]

    sys.path.insert(0, path)
    try:
        ref_model = importlib.import_module('model')
        ref_losses = importlib.import_module('utils.losses')
    finally:
        sys.path.remove(path)
    _REF = (ref_model, ref_losses)
    return _REF


def load_reference_push(path='/root/reference'):
    """Import the reference push module (needs the same stubs)."""
    load_reference(path)
    sys.path.insert(0, path)
    try:
        return importlib.import_module('push')
    finally:
        sys.path.remove(path)


def make_batches(n_steps, batch, C, img, seed=7):
    """Separable synthetic batches (class pattern + noise), pre-generated
    so both runs see bit-identical inputs."""
    g = torch.Generator().manual_seed(seed)
    patterns = torch.randn(C, 3, img, img, generator=g)
    out = []
    for _ in range(n_steps):
        label = torch.randint(0, C, (batch,), generator=g)
        x = 0.4 * torch.randn(batch, 3, img, img, generator=g) + patterns[label]
        out.append((x, label))
    return out


def _build_ref(C, K, d, img, mem, mine, sz_emb, arch):
    ref_model, ref_losses = load_reference()
    torch.manual_seed(0)
    net = ref_model.construct_MGProto(
        arch, pretrained=False, img_size=img, prototype_shape=(C * K, d, 1, 1),
        num_classes=C, add_on_layers_type='regular', sz_embedding=sz_emb,
        mem_capacity=mem, mine_K=mine)
    torch.manual_seed(1)
    aux = ref_losses.Proxy_Anchor(nb_classes=C, sz_embed=sz_emb, mrg=0.1,
                                  beta=32)
    return net, aux


def _build_ours(C, K, d, img, mem, mine, sz_emb, arch):
    from mgproto_amd.model import construct_MGProto
    from mgproto_amd.losses import build_aux_loss
    torch.manual_seed(0)
    net = construct_MGProto(
        arch, pretrained=False, img_size=img, prototype_shape=(C * K, d, 1, 1),
        num_classes=C, add_on_layers_type='regular', sz_embedding=sz_emb,
        mem_capacity=mem, mine_K=mine)
    torch.manual_seed(1)
    aux = build_aux_loss('Proxy_Anchor', nb_classes=C, sz_embed=sz_emb,
                         mrg=0.1, beta=32)
    return net, aux


def _optim(net, aux, proto_lr=3e-3):
    opt = torch.optim.Adam([
        {'params': net.features.parameters(), 'lr': 1e-4, 'weight_decay': 1e-4},
        {'params': net.add_on_layers.parameters(), 'lr': 3e-3, 'weight_decay': 1e-4},
        {'params': aux.parameters(), 'lr': 1e-2, 'weight_decay': 1e-4},
    ])
    return opt


def run_side(net, aux, batches, is_reference, proto_lr=3e-3):
    """One training run, mirroring reference train_and_test._training:10-63
    (joint phase, mining on, EM on once the bank is non-empty)."""
    if is_reference:
        net.prototype_optimizer = torch.optim.Adam(
            [{'params': net.prototype_means, 'lr': proto_lr}])
    else:
        net.prototype_lr = proto_lr
    opt = _optim(net, aux)
    coefs = {'crs_ent': 1.0, 'mine': 0.2, 'aux': 0.5}
    net.train()
    traj = []
    for image, target in batches:
        output, x_aux = net(image, target)
        mine_loss = sum(F.cross_entropy(output[:, :, k], target)
                        for k in range(1, output.shape[2])) \
            / (output.shape[2] - 1)
        ce = F.cross_entropy(output[:, :, 0], target)
        aux_loss = aux(x_aux, target)
        loss = coefs['crs_ent'] * ce + coefs['mine'] * mine_loss \
            + coefs['aux'] * aux_loss
        opt.zero_grad()
        loss.backward()
        opt.step()
        # EM trigger exactly as reference train_and_test.py:61-63
        if net.queue.mem_len.sum() > 0 \
                and int(net.iteration_counter) % net.update_interval == 0:
            net.update_GMM()
        acc = (output[:, :, 0].argmax(1) == target).float().mean()
        traj.append({'loss': float(loss), 'ce': float(ce),
                     'mine': float(mine_loss), 'aux': float(aux_loss),
                     'acc': float(acc)})
    return traj


def run_ab(n_steps=32, batch=16, C=4, K=2, d=16, img=32, mem=8, mine=2,
           sz_emb=8, arch='resnet18'):
    """Returns (ref_traj, ours_traj). Both sides start from the
    REFERENCE's initial weights: ours loads them via load_state_dict —
    the checkpoint-interop contract exercised with a real
    reference-produced state_dict."""
    batches = make_batches(n_steps, batch, C, img)
    ref_net, ref_aux = _build_ref(C, K, d, img, mem, mine, sz_emb, arch)
    our_net, our_aux = _build_ours(C, K, d, img, mem, mine, sz_emb, arch)
    our_net.load_state_dict(ref_net.state_dict())           # strict
    with torch.no_grad():
        our_aux.proxies.copy_(ref_aux.proxies)

    ref_traj = run_side(ref_net, ref_aux, batches, is_reference=True)
    our_traj = run_side(our_net, our_aux, batches, is_reference=False)
    return ref_traj, our_traj


def render_table(ref_traj, our_traj, meta=''):
    lines = ['# Learning-equivalence A/B: reference vs mgproto_amd', '',
             meta, '',
             '| step | ref loss | ours loss | ref CE | ours CE | '
             'ref acc | ours acc |',
             '|---|---|---|---|---|---|---|']
    for i, (r, o) in enumerate(zip(ref_traj, our_traj)):
        lines.append(f"| {i} | {r['loss']:.4f} | {o['loss']:.4f} "
                     f"| {r['ce']:.4f} | {o['ce']:.4f} "
                     f"| {r['acc']:.3f} | {o['acc']:.3f} |")
    dl = [abs(r['loss'] - o['loss']) for r, o in zip(ref_traj, our_traj)]
    lines += ['', f'max |Δloss| = {max(dl):.4f}; final ref loss '
              f"{ref_traj[-1]['loss']:.4f} vs ours {our_traj[-1]['loss']:.4f}"]
    return '\n'.join(lines) + '\n'


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--steps', type=int, default=32)
    ap.add_argument('--batch', type=int, default=16)
    ap.add_argument('--arch', type=str, default='resnet18')
    ap.add_argument('--out', type=str, default='profiles/ab_reference.md')
    args = ap.parse_args()
    torch.manual_seed(0)
    ref_traj, our_traj = run_ab(n_steps=args.steps, batch=args.batch,
                                arch=args.arch)
    meta = (f'arch={args.arch}, steps={args.steps}, batch={args.batch}, '
            f'C=4 K=2 d=16 img=32 (CPU fp32; identical weights/batches; '
            f'joint phase, mining+EM on)')
    md = render_table(ref_traj, our_traj, meta)
    os.makedirs(os.path.dirname(args.out), exist_ok=True)
    with open(args.out, 'w') as f:
        f.write(md)
    print(md)


if __name__ == '__main__':
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    main()
