#!/usr/bin/env python3
"""Training driver — the reference ``main.py`` rebuilt for MI355X.

Single GPU:  python train.py -arch resnet34 -dataset CUB
8-GPU node:  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
                 --master-addr 127.0.0.1 train.py -arch resnet50 ...

Differences from the reference driver (main.py):
* one process per GPU over RCCL instead of DataParallel (main.py:184);
* real input pipeline (num_workers > 0; the reference ran the loader on the
  main thread, main.py:94);
* full checkpoint/resume (the reference can only save);
* works on synthetic data out of the box (-dataset synthetic) since this
  environment has no datasets.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch
from torch.utils.data import DataLoader

from mgproto_amd import settings
from mgproto_amd.settings import Settings
from mgproto_amd.model import construct_MGProto
from mgproto_amd.losses import build_aux_loss
from mgproto_amd.engine import trainer as tnt
from mgproto_amd.engine.push import push_prototypes
from mgproto_amd.engine.trainer import EMRunner
from mgproto_amd.parallel import Comm, BucketedGradReducer, make_dp_correct
from mgproto_amd.utils import makedir, datestr, create_logger, MetricsLogger
from mgproto_amd.utils.checkpoint import (save_model_w_condition,
                                          save_train_state, load_train_state)
from mgproto_amd.data.synthetic import SyntheticImages


def build_loaders(cfg: Settings, comm):
    """Train/push/test/ood loaders. Real image folders when the dirs exist,
    synthetic otherwise."""
    from mgproto_amd.data import loaders as L
    world = comm.world_size if comm else 1
    rank = comm.rank if comm else 0
    if os.path.isdir(cfg.train_dir):
        return L.build_image_loaders(cfg, world=world, rank=rank)
    # synthetic fallback (no datasets in this environment)
    n = 32 if os.environ.get('MGPROTO_TINY_TEST') == '1' else 1600
    train_ds = SyntheticImages(n, cfg.num_classes, cfg.img_size)
    push_ds = SyntheticImages(n, cfg.num_classes, cfg.img_size, normalize=False)
    test_ds = SyntheticImages(n // 4, cfg.num_classes, cfg.img_size, seed=1)
    ood1 = SyntheticImages(n // 8, cfg.num_classes, cfg.img_size, seed=2)
    ood2 = SyntheticImages(n // 8, cfg.num_classes, cfg.img_size, seed=3)

    def shard(ds):
        if world > 1:
            from torch.utils.data import Subset
            return Subset(ds, list(range(rank, len(ds), world)))
        return ds

    def collate(batch):
        return (torch.stack([b[0] for b in batch]),
                torch.tensor([b[1] for b in batch]),
                torch.tensor([b[2] for b in batch]))

    mk = lambda ds, bs, sampler=None: DataLoader(  # noqa: E731
        ds, batch_size=bs, num_workers=cfg.num_workers,
        collate_fn=collate, drop_last=False, sampler=sampler)
    from mgproto_amd.data.loaders import make_train_sampler
    train_sampler = make_train_sampler(train_ds, world, rank)
    return (mk(train_ds, cfg.train_batch_size, sampler=train_sampler),
            mk(shard(push_ds), cfg.train_push_batch_size),
            mk(shard(test_ds), cfg.test_batch_size),
            mk(shard(ood1), cfg.test_batch_size),
            mk(shard(ood2), cfg.test_batch_size))


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument('-gpuid', type=str, default='0')      # reference compat
    parser.add_argument('-dataset', type=str, default='CUB')
    parser.add_argument('-arch', type=str, default='resnet34')
    parser.add_argument('-aux_loss', type=str, default='Proxy_Anchor')
    parser.add_argument('-aux_emb_sz', type=int, default=32)
    parser.add_argument('-mem_sz', type=int, default=800)
    parser.add_argument('-mine_level', type=int, default=20)
    parser.add_argument('--epochs', type=int, default=None)
    parser.add_argument('--resume', type=str, default=None)
    parser.add_argument('--out', type=str, default=None)
    parser.add_argument('--ood-eval', action='store_true')
    parser.add_argument('--prefetch', action='store_true',
                        help='stage next-batch H2D copies on a side stream '
                             '(real-data runs; the synthetic bench path is '
                             'already device-resident)')
    parser.add_argument('--addon', type=str, default=None)
    parser.add_argument('--no-auto-resume', action='store_true',
                        help='do not resume from an existing latest.pth / '
                             'preempt.pth in --out (elastic-restart default: '
                             'resume, so a torchrun relaunch after preemption '
                             'continues the run)')
    parser.add_argument('--seed', type=int, default=None,
                        help='deterministic init/data seeding (the reference '
                             'ships seeding commented out, main.py:45-49)')
    parser.add_argument('--faithful-aug', action='store_true',
                        help='use the reference\'s exact 4-pass augmentation '
                             'chain (main.py:98-104) instead of the fused '
                             'one-homography approximation — pick this for '
                             'training-parity comparisons')
    args = parser.parse_args()

    cfg = Settings(base_architecture=args.arch, aux_loss=args.aux_loss,
                   sz_embedding=args.aux_emb_sz, mem_capacity=args.mem_sz,
                   mine_K=args.mine_level,
                   fast_augment=not args.faithful_aug)
    if args.addon:
        cfg.add_on_layers_type = args.addon
    if os.environ.get('MGPROTO_TINY_TEST') == '1':
        # CI-sized config: small synthetic problem, CPU-friendly
        cfg.img_size = 64
        cfg.num_classes = 10
        cfg.prototype_shape = (30, 32, 1, 1)
        cfg.add_on_layers_type = 'regular'
        cfg.train_batch_size = cfg.test_batch_size = 8
        cfg.train_push_batch_size = 8
        cfg.num_workers = 0
        cfg.mine_start = 0
        cfg.updateGMM_start = 0
        cfg.push_start = 0
        cfg.push_epochs = [0]
    if args.epochs is not None:
        cfg.num_train_epochs = args.epochs
        cfg.push_epochs = [i for i in range(args.epochs) if i % 10 == 0]

    if args.seed is not None:
        import random as _random
        _random.seed(args.seed)
        torch.manual_seed(args.seed)
        if torch.cuda.is_available():
            torch.cuda.manual_seed_all(args.seed)

    comm = Comm() if int(os.environ.get('WORLD_SIZE', '1')) > 1 else None
    rank = comm.rank if comm else 0
    device = (comm.device if comm
              else (torch.device('cuda', 0) if torch.cuda.is_available()
                    else torch.device('cpu')))
    if device.type == 'cuda':
        torch.cuda.set_device(device)
        from mgproto_amd.utils.helpers import setup_miopen_db
        setup_miopen_db()
        torch.backends.cudnn.benchmark = True

    model_dir = args.out or 'saved_models/{}/{}/'.format(args.arch, datestr())
    if comm is not None and comm.is_distributed:
        # ranks must agree on the run dir (datestr() can straddle a minute
        # boundary): rank 0's choice wins
        buf = torch.zeros(512, dtype=torch.uint8, device=comm.device)
        if rank == 0:
            raw = model_dir.encode()[:512]
            buf[:len(raw)] = torch.tensor(list(raw), dtype=torch.uint8,
                                          device=comm.device)
        comm.broadcast(buf)
        model_dir = bytes(b for b in buf.cpu().tolist() if b).decode()
    if rank == 0:
        makedir(model_dir)
        makedir(os.path.join(model_dir, 'img'))
    if comm:
        comm.barrier()
    log, logclose = (create_logger(os.path.join(model_dir, 'train.log'))
                     if rank == 0 else (lambda s: None, lambda: None))
    metrics = MetricsLogger(os.path.join(model_dir, 'metrics.jsonl'), rank=rank)

    loaders = build_loaders(cfg, comm)
    train_loader, push_loader, test_loader, ood1_loader, ood2_loader = loaders
    if args.prefetch and device.type == 'cuda':
        from mgproto_amd.data.prefetch import DevicePrefetcher
        train_loader = DevicePrefetcher(train_loader, device)

    ppnet = construct_MGProto(
        base_architecture=args.arch, pretrained=True, img_size=cfg.img_size,
        prototype_shape=cfg.prototype_shape, num_classes=cfg.num_classes,
        prototype_activation_function=cfg.prototype_activation_function,
        add_on_layers_type=cfg.add_on_layers_type,
        sz_embedding=args.aux_emb_sz, mem_capacity=args.mem_sz,
        mine_K=args.mine_level).to(device)
    if device.type == 'cuda' and cfg.channels_last:
        ppnet.features = ppnet.features.to(memory_format=torch.channels_last)

    aux_criterion = build_aux_loss(args.aux_loss, nb_classes=cfg.num_classes,
                                   sz_embed=args.aux_emb_sz, mrg=0.1,
                                   beta=32).to(device)

    from mgproto_amd.engine.graphstep import GraphedStep, graphs_enabled
    world = comm.world_size if comm else 1
    use_graph = cfg.hip_graph and graphs_enabled(device, world)

    def _adam(groups, capturable):
        # fused+capturable Adam keeps the whole optimizer step on device
        # (required inside a hipGraph; harmless outside one)
        try:
            return torch.optim.Adam(groups, fused=True, capturable=capturable)
        except (RuntimeError, TypeError, ValueError):
            return torch.optim.Adam(groups, capturable=capturable)

    joint_optimizer = _adam([
        {'params': ppnet.features.parameters(),
         'lr': cfg.joint_optimizer_lrs['features'], 'weight_decay': 1e-4},
        {'params': ppnet.add_on_layers.parameters(),
         'lr': cfg.joint_optimizer_lrs['add_on_layers'], 'weight_decay': 1e-4},
        {'params': aux_criterion.parameters(),
         'lr': cfg.joint_optimizer_lrs['features'] * 100, 'weight_decay': 1e-4},
    ], use_graph)
    joint_lr_scheduler = torch.optim.lr_scheduler.StepLR(
        joint_optimizer, step_size=1, gamma=cfg.joint_lr_gamma)
    warm_optimizer = _adam([
        {'params': ppnet.add_on_layers.parameters(),
         'lr': cfg.warm_optimizer_lrs['add_on_layers'], 'weight_decay': 1e-4},
        {'params': aux_criterion.parameters(),
         'lr': cfg.joint_optimizer_lrs['features'] * 100, 'weight_decay': 1e-4},
    ], use_graph)
    ppnet.prototype_lr = cfg.joint_optimizer_lrs['prototype_vectors']

    graph_step = None
    if use_graph:
        graph_step = GraphedStep(ppnet, aux_criterion, cfg.coefs, device,
                                 cfg.train_batch_size, cfg.img_size,
                                 amp_dtype=cfg.amp_dtype,
                                 channels_last=cfg.channels_last)
        log('hipGraph training step: enabled')

    reducer = None
    if comm is not None and comm.is_distributed:
        ppnet = make_dp_correct(ppnet, comm, cfg.train_batch_size)
        comm.broadcast_module(aux_criterion)
        reducer = BucketedGradReducer([ppnet, aux_criterion], comm,
                                      bucket_mb=cfg.grad_bucket_mb)

    em_runner = EMRunner(ppnet, use_stream=(device.type == 'cuda'
                                            and cfg.em_stream))
    start_epoch = 0
    resume_path = args.resume
    if resume_path is None and not args.no_auto_resume:
        # elastic restart: a relaunch after preemption/crash picks up the
        # newest checkpoint in the run dir automatically
        cands = [os.path.join(model_dir, f)
                 for f in ('latest.pth', 'preempt.pth')]
        cands = [p for p in cands if os.path.isfile(p)]
        if cands:
            resume_path = max(cands, key=os.path.getmtime)
    if resume_path:
        state = load_train_state(resume_path, ppnet,
                                 {'joint': joint_optimizer, 'warm': warm_optimizer},
                                 {'joint_lr': joint_lr_scheduler},
                                 map_location=device)
        start_epoch = state['epoch'] + 1
        log(f'resumed from {resume_path} at epoch {start_epoch}')

    # failure handling: checkpoint on SIGTERM/SIGUSR1 (preemption-safe;
    # the reference loses everything past the last conditional save)
    import signal

    def _save_now(signum, frame):
        if rank == 0:
            save_train_state(os.path.join(model_dir, 'preempt.pth'), ppnet,
                             {'joint': joint_optimizer, 'warm': warm_optimizer},
                             {'joint_lr': joint_lr_scheduler},
                             epoch=_save_now.epoch,
                             extra={'reason': f'signal {signum}'})
            log(f'checkpointed on signal {signum}')
    _save_now.epoch = start_epoch - 1
    for sig in (signal.SIGTERM, signal.SIGUSR1):
        try:
            signal.signal(sig, _save_now)
        except (ValueError, OSError):
            pass

    log('start training')
    decay_epochs = cfg.lr_decay_epochs()
    epoch = start_epoch
    last_phase = None
    for epoch in range(start_epoch, cfg.num_train_epochs):
        _save_now.epoch = epoch
        # global per-epoch reshuffle (DistributedSampler contract); the
        # prefetcher wrapper exposes the underlying loader's sampler
        sampler = getattr(getattr(train_loader, 'loader', train_loader),
                          'sampler', None)
        if hasattr(sampler, 'set_epoch'):
            sampler.set_epoch(epoch)
        log('epoch: \t{0}'.format(epoch))
        use_mining = epoch >= cfg.mine_start
        update_GMM = (epoch >= cfg.updateGMM_start
                      and bool((ppnet.queue.mem_len == ppnet.capacity_pc).all()))
        log('use mining: \t{0}'.format(use_mining))
        log('update GMM: \t{0}'.format(update_GMM))

        kw = dict(device=device, amp_dtype=cfg.amp_dtype, comm=comm,
                  metrics=metrics)
        phase = 'warm' if epoch < cfg.num_warm_epochs else 'joint'
        if phase == 'warm':
            tnt.warm_only(ppnet, log=log)
            optimizer = warm_optimizer
        else:
            tnt.joint(ppnet, log=log)
            if epoch in decay_epochs:
                joint_lr_scheduler.step()
            optimizer = joint_optimizer
        if reducer is not None and phase != last_phase:
            # bucket membership follows requires_grad: rebuild on phase
            # flips so warm-phase reduces overlap with backward too
            reducer.rebuild()
        last_phase = phase
        _, train_results = tnt.train(
            ppnet, train_loader, optimizer, aux_criterion=aux_criterion,
            use_mine=use_mining, update_GMM=update_GMM, coefs=cfg.coefs,
            log=log, em_runner=em_runner, reducer=reducer,
            graph_step=graph_step, **kw)

        if args.ood_eval:
            accu, _ = tnt.test(ppnet, (test_loader, ood1_loader, ood2_loader),
                               log=log, ood=True, **kw)
        else:
            accu, _ = tnt.test(ppnet, (test_loader,), log=log, **kw)
        if rank == 0:
            save_model_w_condition(ppnet, model_dir, str(epoch) + 'nopush',
                                   accu, 0.00, log=log)
            save_train_state(os.path.join(model_dir, 'latest.pth'), ppnet,
                             {'joint': joint_optimizer, 'warm': warm_optimizer},
                             {'joint_lr': joint_lr_scheduler}, epoch)
        metrics.log({'epoch': epoch,
                     'lr': joint_optimizer.param_groups[0]['lr'],
                     'update_GMM': update_GMM * 1.0,
                     'use_mining': use_mining * 1.0})

        if epoch >= cfg.push_start and epoch in cfg.push_epochs:
            push_prototypes(
                push_loader, ppnet, class_specific=True,
                preprocess_input_function=_preprocess(),
                root_dir_for_saving_prototypes=os.path.join(model_dir, 'img'),
                epoch_number=epoch, prototype_img_filename_prefix='prototype-img',
                log=log, comm=comm, device=device)
            accu, _ = tnt.test(ppnet, (test_loader,), log=log, **kw)
            if rank == 0:
                save_model_w_condition(ppnet, model_dir, str(epoch) + 'push',
                                       accu, 0.00, log=log)

    # final pruning (reference main.py:285)
    ppnet.prune_prototypes_topM(top_M=min(8, ppnet.num_prototypes_per_class))
    accu, _ = tnt.test(ppnet, (test_loader,), log=log, device=device,
                       amp_dtype=cfg.amp_dtype, comm=comm)
    if rank == 0:
        save_model_w_condition(ppnet, model_dir, str(epoch) + 'prune', accu,
                               0.00, log=log)
    metrics.close()
    logclose()


def _preprocess():
    from mgproto_amd.data.preprocess import preprocess_input_function
    return preprocess_input_function


if __name__ == '__main__':
    main()
