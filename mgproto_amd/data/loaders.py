"""DataLoader construction for the real-image pipeline.

Reproduces the reference's five loaders (train / push / test / ood1 / ood2,
main.py:96-163) with the same transform stacks, but with worker processes
and distributed sharding (the reference ran num_workers=0 on the main
thread — SURVEY.md hard part #5)."""

from typing import Tuple

import torch
from torch.utils.data import DataLoader, Subset
from torch.utils.data.distributed import DistributedSampler

from . import transforms as T
from .folder import ImageFolder
from .preprocess import mean, std


def _collate(batch):
    return (torch.stack([b[0] for b in batch]),
            torch.tensor([b[1] for b in batch]),
            torch.tensor([b[2] for b in batch]))


def _shard(ds, world, rank):
    if world > 1:
        return Subset(ds, list(range(rank, len(ds), world)))
    return ds


def make_train_sampler(ds, world: int, rank: int, seed: int = 0):
    """Globally-reshuffled, padded train sharding.

    Every epoch all ranks draw from ONE global permutation (call
    ``sampler.set_epoch(epoch)`` — the trainer does), and shards are padded
    to equal length so every rank issues the same number of gradient
    all-reduces per epoch. A static ``range(rank, N, world)`` shard would
    (a) freeze which samples a rank ever sees — the reference's
    DataParallel saw globally shuffled batches every epoch
    (reference main.py:96-104) — and (b) let per-rank batch counts differ
    by one, desynchronizing the reducer's RCCL collectives."""
    if world <= 1:
        return None
    return DistributedSampler(ds, num_replicas=world, rank=rank,
                              shuffle=True, seed=seed, drop_last=False)


def build_image_loaders(cfg, world: int = 1, rank: int = 0,
                        fast_augment: bool = None,
                        scaled_decode: bool = None) -> Tuple:
    img_size = cfg.img_size
    import os
    if scaled_decode is None:
        scaled_decode = os.environ.get('MGPROTO_SCALED_DECODE', '0') == '1'
    if fast_augment is None:
        fast_augment = getattr(cfg, 'fast_augment', True)
    normalize = T.Normalize(mean=mean, std=std)

    if fast_augment:
        # one-homography fused pipeline (~4x the faithful chain's
        # throughput; see FusedTrainTransform) — same parameters as the
        # reference stack (main.py:98-104)
        train_tf = T.FusedTrainTransform(img_size, scale=(0.60, 1.0),
                                         normalize=normalize)
    else:
        train_tf = T.Compose([
            T.RandomPerspective(distortion_scale=0.2, p=0.5),
            T.ColorJitter((0.6, 1.4), (0.6, 1.4), (0.6, 1.4), (-0.02, 0.02)),
            T.RandomHorizontalFlip(),
            T.RandomAffine(degrees=25, shear=(-15, 15), translate=[0.05, 0.05]),
            T.RandomResizedCrop(size=(img_size, img_size), scale=(0.60, 1.0)),
            T.ToTensor(),
            normalize,
        ])
    push_tf = T.Compose([
        T.Resize(size=(img_size, img_size)),
        T.ToTensor(),
    ])
    test_tf = T.Compose([
        T.Resize(img_size + 32),
        T.CenterCrop(img_size),
        T.ToTensor(),
        normalize,
    ])
    ood_tf = T.Compose([
        T.Resize(size=(img_size, img_size)),
        T.ToTensor(),
        normalize,
    ])

    # Train only: 2x oversampling headroom is kept (min crop scale 0.6 of a
    # 2*S decode still exceeds S); push/test keep full-fidelity decodes.
    train_ds = ImageFolder(cfg.train_dir, train_tf,
                           decode_size=2 * img_size if scaled_decode else None)
    push_ds = ImageFolder(cfg.train_push_dir, push_tf)
    test_ds = ImageFolder(cfg.test_dir, test_tf)

    def mk(ds, bs, shuffle=False, sampler=None):
        return DataLoader(ds, batch_size=bs,
                          shuffle=shuffle if sampler is None else False,
                          sampler=sampler,
                          num_workers=cfg.num_workers, pin_memory=True,
                          collate_fn=_collate, persistent_workers=cfg.num_workers > 0)

    train_sampler = make_train_sampler(train_ds, world, rank)
    train_loader = mk(train_ds, cfg.train_batch_size,
                      shuffle=True, sampler=train_sampler)
    # push/test shards stay static stride shards: they run under no_grad
    # with a single post-loop collective, so uneven batch counts are safe,
    # and padding would double-count images in the metrics
    push_loader = mk(_shard(push_ds, world, rank), cfg.train_push_batch_size)
    test_loader = mk(_shard(test_ds, world, rank), cfg.test_batch_size)

    ood_loaders = []
    for d in (cfg.test_dir_ood1, cfg.test_dir_ood2):
        if os.path.isdir(d):
            ood_loaders.append(mk(_shard(ImageFolder(d, ood_tf), world, rank),
                                  cfg.test_batch_size))
        else:
            ood_loaders.append(None)
    return (train_loader, push_loader, test_loader,
            ood_loaders[0], ood_loaders[1])
