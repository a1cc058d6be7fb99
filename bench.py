#!/usr/bin/env python3
"""Flagship training-step benchmark (driver contract).

Measures the BASELINE.json headline metric: train images/sec (whole job) on
the CUB-200 ResNet-50 MGProto config — 224x224 synthetic data, 200 classes,
2000 prototypes (d=64), 28x28 latent grid (the reference's R50 headline
checkpoint config), batch 80 per GPU, bf16 backbone autocast,
fp32 prototype math, mining + memory-bank enqueue + EM all active
(steady-state joint-phase training step).

Single GPU:   python bench.py --steps 20 --warmup 5
Multi-GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""

import argparse
import json
import os
import time

import torch
import torch.nn.functional as F


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--gpus', type=int, default=1)
    ap.add_argument('--steps', type=int, default=100)
    ap.add_argument('--warmup', type=int, default=10)
    ap.add_argument('--batch', type=int, default=80)
    ap.add_argument('--arch', type=str, default='resnet50')
    ap.add_argument('--addon', type=str, default='regular_upsample',
                    help='regular (14x14 grid) | regular_upsample (28x28)')
    ap.add_argument('--classes', type=int, default=200)
    ap.add_argument('--proto-dim', type=int, default=64)
    ap.add_argument('--proto-per-class', type=int, default=10)
    ap.add_argument('--mine', type=int, default=20)
    ap.add_argument('--mem', type=int, default=800)
    ap.add_argument('--img', type=int, default=224)
    ap.add_argument('--no-em', action='store_true')
    ap.add_argument('--no-graph', action='store_true',
                    help='disable hipGraph capture of the training step')
    ap.add_argument('--eager', action='store_true',
                    help='allow the PyTorch fallback for the prototype ops')
    ap.add_argument('--preset', type=str, default=None,
                    choices=['cub-r50', 'cars-densenet161', 'ood-resnet152',
                             'pets-vgg19'],
                    help='BASELINE.json config presets (2-5); flags still '
                         'override')
    # presets (BASELINE.json configs): apply before final parse so explicit
    # flags win
    import sys as _sys
    pre, _ = ap.parse_known_args()
    presets = {
        'cars-densenet161': dict(arch='densenet161', classes=196,
                                 addon='regular'),
        'ood-resnet152': dict(arch='resnet152', classes=200, addon='regular'),
        'pets-vgg19': dict(arch='vgg19', classes=37, batch=256, mem=4000,
                           addon='regular'),
    }
    if pre.preset in presets:
        explicit = {a.lstrip('-').replace('-', '_').split('=')[0]
                    for a in _sys.argv[1:] if a.startswith('--')}
        for k, v in presets[pre.preset].items():
            if k not in explicit:
                ap.set_defaults(**{k: v})
    args = ap.parse_args()

    if args.eager:
        os.environ['MGPROTO_EAGER_FALLBACK'] = '1'

    from mgproto_amd.model import construct_MGProto
    from mgproto_amd.data.synthetic import DeviceBatchPool
    from mgproto_amd.losses import build_aux_loss
    from mgproto_amd.parallel import Comm, BucketedGradReducer, make_dp_correct
    from mgproto_amd.engine.trainer import EMRunner

    from mgproto_amd.utils.helpers import setup_miopen_db
    setup_miopen_db()
    # MIOpen exhaustive find for the (static) conv shapes: the search cost
    # lands in warmup; without it MIOpen's fallback solvers dominate the step
    # (conv bwd-weight measured at 74% of GPU time under FIND_MODE=FAST)
    torch.backends.cudnn.benchmark = True

    world = int(os.environ.get('WORLD_SIZE', '1'))
    comm = Comm() if world > 1 else None
    device = comm.device if comm is not None else (
        torch.device('cuda', 0) if torch.cuda.is_available() else torch.device('cpu'))
    if device.type == 'cuda':
        torch.cuda.set_device(device)
    rank = comm.rank if comm is not None else 0

    C, K, d = args.classes, args.proto_per_class, args.proto_dim
    torch.manual_seed(1234)
    model = construct_MGProto(args.arch, pretrained=False, img_size=args.img,
                              prototype_shape=(C * K, d, 1, 1), num_classes=C,
                              add_on_layers_type=args.addon, sz_embedding=32,
                              mem_capacity=args.mem, mine_K=args.mine)
    model = model.to(device)
    if device.type == 'cuda':
        model.features = model.features.to(memory_format=torch.channels_last)

    # whole-step hipGraph capture: single-node-per-rank eager graphs replace
    # the reference's (nonexistent) tracing compiler — cuts ~2k kernel-launch
    # overheads per step.  Multi-rank capture (RCCL collectives inside the
    # graph) is opt-in via MGPROTO_GRAPH_DIST=1 pending 8-GPU validation.
    use_graph = (device.type == 'cuda') and not args.no_graph and (
        world == 1 or os.environ.get('MGPROTO_GRAPH_DIST') == '1')

    aux = build_aux_loss('Proxy_Anchor', nb_classes=C, sz_embed=32).to(device)
    groups = [
        {'params': model.features.parameters(), 'lr': 1e-4, 'weight_decay': 1e-4},
        {'params': model.add_on_layers.parameters(), 'lr': 3e-3, 'weight_decay': 1e-4},
        {'params': model.embedding.parameters(), 'lr': 3e-3, 'weight_decay': 1e-4},
        {'params': aux.parameters(), 'lr': 1e-2, 'weight_decay': 1e-4},
    ]
    try:
        opt = torch.optim.Adam(groups, fused=True, capturable=use_graph)
    except (RuntimeError, TypeError, ValueError):
        opt = torch.optim.Adam(groups, capturable=use_graph)

    reducer = None
    if comm is not None and comm.is_distributed:
        model = make_dp_correct(model, comm, args.batch)
        comm.broadcast_module(aux)
        reducer = BucketedGradReducer([model, aux], comm, bucket_mb=50)

    # steady-state: memory bank pre-filled so the EM path is active from the
    # first measured step (as in epochs >= updateGMM_start of real training)
    with torch.no_grad():
        mem = F.normalize(torch.randn(C * args.mem, d, device=device), dim=1)
        labels = torch.arange(C, device=device).repeat_interleave(args.mem)
        model.queue.push(mem, labels)
        model.memory_updated_cls[:] = True

    pool = DeviceBatchPool(args.batch, C, args.img, device=device, pool=4,
                           seed=17 + rank,
                           channels_last=(device.type == 'cuda'))
    em = EMRunner(model, use_stream=(device.type == 'cuda' and not use_graph))
    coefs = {'crs_ent': 1.0, 'mine': 0.2, 'aux': 0.5}
    use_em = not args.no_em
    amp = torch.autocast(device_type='cuda', dtype=torch.bfloat16) \
        if device.type == 'cuda' else None

    def compute(image, target):
        if amp is not None:
            with amp:
                output, x_aux = model(image, target)
        else:
            output, x_aux = model(image, target)
        output = output.float()
        mine_loss = sum(F.cross_entropy(output[:, :, k], target)
                        for k in range(1, output.shape[2])) / (output.shape[2] - 1)
        ce = F.cross_entropy(output[:, :, 0], target)
        aux_loss = aux(x_aux.float(), target)
        loss = coefs['crs_ent'] * ce + coefs['mine'] * mine_loss + coefs['aux'] * aux_loss
        if reducer is not None:
            reducer.prepare()
        opt.zero_grad(set_to_none=True)
        loss.backward()
        if reducer is not None:
            reducer.finalize()
        opt.step()
        if use_em:
            if em.use_stream:
                em.run()
            else:
                model.update_GMM()

    def step():
        image, target = pool.next()
        em.sync()
        compute(image, target)

    model.train()
    graph = None
    if use_graph:
        # the SAME whole-step capture train.py runs (engine/graphstep.py) —
        # the bench measures the production path, not a bench-only one
        from mgproto_amd.engine.graphstep import GraphedStep
        gs = GraphedStep(model, aux, coefs, device, args.batch, args.img,
                         amp_dtype='bf16', channels_last=True)

        def step():  # noqa: F811
            image, target = pool.next()
            if gs.matches(image, opt, True, use_em):
                gs.step(image, target, opt, reducer=reducer,
                        em_active=use_em, use_mine=True)
            else:
                compute(image, target)
        # prime: eager warmups + capture happen inside gs.step; make sure
        # they are done before the timed region regardless of --warmup
        for _ in range(gs.warmup_steps + 1):
            step()
        graph = gs.graph
        if graph is None:
            use_graph = False

    for _ in range(args.warmup):
        step()

    if comm is not None:
        comm.barrier()
    if device.type == 'cuda':
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    em.sync()
    if device.type == 'cuda':
        torch.cuda.synchronize()
    if comm is not None:
        comm.barrier()
    elapsed = time.perf_counter() - t0

    # MAX elapsed over ranks -> whole-job throughput
    if comm is not None:
        t = torch.tensor([elapsed], device=device)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t)

    n_gpus = world if world > 1 else 1
    total_images = n_gpus * args.batch * args.steps
    ips = total_images / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        # stride-16 trunk; the upsample add-on doubles the latent grid
        grid = (args.img // 16) * (2 if args.addon == 'regular_upsample' else 1)
        print(json.dumps({
            'metric': 'train_images_per_sec',
            'value': round(ips, 2),
            'unit': 'images/s',
            'n_gpus': n_gpus,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': round(ms_per_step, 3),
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': None,
            'dtype': 'bf16' if amp is not None else 'fp32',
            'data': 'synthetic',
            'config': {
                'model': f'{args.arch}-mgproto',
                'global_batch': n_gpus * args.batch,
                'img_size': args.img,
                'latent_grid': f'{grid}x{grid}',
                'num_classes': C,
                'prototypes': C * K,
                'proto_dim': d,
                'mine_T': args.mine,
                'mem_capacity': args.mem,
                'em_active': use_em,
                'hip_graph': bool(graph is not None),
                'parallelism': f'dp{n_gpus}',
            },
        }))


if __name__ == '__main__':
    main()
