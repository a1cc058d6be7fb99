"""Round-2 correctness fixes (ADVICE.md + VERDICT.md #5).

* distributed push candidates staged through comm.device (ADVICE high);
* padded, globally-reshuffled train sampler -> equal per-rank step counts
  and a real per-epoch reshuffle (ADVICE medium, VERDICT missing #4);
* EM Adam state round-trips through save_train_state (ADVICE low);
* reducer rebuild across warm->joint requires_grad flips, fuzzed over
  rank counts (VERDICT weak #5).
"""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _run_workers(fn, world=2, extra=()):
    ctx = mp.get_context('spawn')
    port = str(29810 + (os.getpid() + hash(fn.__name__)) % 150)
    procs = []
    q = ctx.SimpleQueue()
    for rank in range(world):
        p = ctx.Process(target=_worker_entry,
                        args=(fn.__name__, rank, world, port, q, extra))
        p.start()
        procs.append(p)
    results = {}
    for _ in range(world):
        rank, payload = q.get()
        if isinstance(payload, str) and payload.startswith('ERROR'):
            for p in procs:
                p.terminate()
            pytest.fail(f'rank {rank}: {payload}')
        results[rank] = payload
    for p in procs:
        p.join(timeout=60)
    return results


def _worker_entry(fn_name, rank, world, port, q, extra):
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = port
    torch.manual_seed(100 + rank)
    try:
        payload = globals()[fn_name](rank, world, *extra)
        q.put((rank, payload))
    except Exception:  # noqa: BLE001
        import traceback
        q.put((rank, 'ERROR ' + traceback.format_exc()))
        raise
    finally:
        import torch.distributed as dist
        if dist.is_initialized():
            dist.destroy_process_group()


# ------------------------------------------------ train sampler semantics

def _sampler_epoch_indices(n, world, epoch, seed=0):
    from mgproto_amd.data.loaders import make_train_sampler

    class _Dummy(torch.utils.data.Dataset):
        def __len__(self):
            return n

        def __getitem__(self, i):
            return i

    per_rank = []
    for rank in range(world):
        s = make_train_sampler(_Dummy(), world, rank, seed=seed)
        s.set_epoch(epoch)
        per_rank.append(list(iter(s)))
    return per_rank


def test_train_sampler_equal_lengths_when_uneven():
    # N=13, world=4: a stride shard would give ranks 4/3/3/3 samples; the
    # padded sampler gives every rank the same count (ADVICE medium: equal
    # per-rank batch counts keep RCCL collectives in lockstep)
    per_rank = _sampler_epoch_indices(13, 4, epoch=0)
    lens = {len(ix) for ix in per_rank}
    assert lens == {4}, lens
    # union covers the whole dataset (padding duplicates, never drops)
    union = set()
    for ix in per_rank:
        union.update(ix)
    assert union == set(range(13))


def test_train_sampler_reshuffles_globally_each_epoch():
    e0 = _sampler_epoch_indices(64, 2, epoch=0)
    e1 = _sampler_epoch_indices(64, 2, epoch=1)
    # different epoch -> different global permutation
    assert e0 != e1
    # a given rank's shard CHANGES across epochs (the old static stride
    # shard froze rank membership forever)
    assert set(e0[0]) != set(e1[0])
    # same epoch is deterministic
    assert e0 == _sampler_epoch_indices(64, 2, epoch=0)


def test_image_loader_uses_padded_sampler(tmp_path):
    # uneven synthetic image tree: 2 classes x 5 images = 10, world=4
    from PIL import Image
    for c in range(2):
        d = tmp_path / f'class_{c}'
        d.mkdir()
        for i in range(5):
            Image.new('RGB', (32, 32), (c * 100, i * 20, 0)).save(
                d / f'im{i}.jpg')

    from mgproto_amd.settings import Settings
    counts = []
    for rank in range(4):
        cfg = Settings(img_size=32, num_workers=0, train_batch_size=3)
        cfg.train_dir = str(tmp_path)
        cfg.train_push_dir = str(tmp_path)
        cfg.test_dir = str(tmp_path)
        from mgproto_amd.data.loaders import build_image_loaders
        train_loader, *_ = build_image_loaders(cfg, world=4, rank=rank)
        counts.append(len(train_loader))
    assert len(set(counts)) == 1, counts


# ------------------------------------------------ push gather device path

def test_gather_push_candidates_returns_host_tensors():
    _run_workers(impl_gather_push_candidates, world=2)


def impl_gather_push_candidates(rank, world):
    from mgproto_amd.parallel import Comm
    from mgproto_amd.parallel.state_sync import gather_push_candidates
    comm = Comm(backend='gloo')
    m = 3 + rank
    dists = torch.arange(m, dtype=torch.float32) + rank * 10
    meta = torch.arange(m * 4, dtype=torch.int64).view(m, 4) + rank * 100
    d_all, m_all = gather_push_candidates(comm, dists, meta)
    # results come back on the host (the greedy merge is host-side numpy)
    assert d_all.device.type == 'cpu' and m_all.device.type == 'cpu'
    assert d_all.numel() == 3 + 4  # rank shards concatenated in rank order
    assert m_all.shape == (7, 4)
    assert torch.equal(d_all[:3], torch.arange(3, dtype=torch.float32))
    assert torch.equal(d_all[3:], torch.arange(4, dtype=torch.float32) + 10)
    return True


# ------------------------------------------------ EM Adam resume state

def test_em_adam_state_survives_resume(tmp_path):
    from mgproto_amd.model import construct_MGProto
    from mgproto_amd.utils.checkpoint import save_train_state, load_train_state
    torch.manual_seed(0)
    m1 = construct_MGProto('resnet18', pretrained=False, img_size=64,
                           prototype_shape=(20, 16, 1, 1), num_classes=5,
                           add_on_layers_type='regular', sz_embedding=8,
                           mem_capacity=4, mine_K=2)
    with torch.no_grad():
        m1._em_exp_avg.normal_()
        m1._em_exp_avg_sq.uniform_(0, 1)
        m1._em_step.fill_(7)
    path = str(tmp_path / 'latest.pth')
    save_train_state(path, m1, {}, {}, epoch=3)

    torch.manual_seed(1)
    m2 = construct_MGProto('resnet18', pretrained=False, img_size=64,
                           prototype_shape=(20, 16, 1, 1), num_classes=5,
                           add_on_layers_type='regular', sz_embedding=8,
                           mem_capacity=4, mine_K=2)
    load_train_state(path, m2)
    assert torch.equal(m2._em_exp_avg, m1._em_exp_avg)
    assert torch.equal(m2._em_exp_avg_sq, m1._em_exp_avg_sq)
    assert torch.equal(m2._em_step, m1._em_step)
    # ... while the exported reference-layout state_dict stays clean
    assert not any(k.startswith('_em_') for k in m1.state_dict())


# ------------------------------------------------ gmm dispatch equivalence

def test_gmm_uniform_and_general_paths_agree_cpu():
    """The uniform-sigma reduced path and the general [x,x^2] path are
    different code (and different kernels on GPU); on identical inputs
    they must produce the same log-probs and feature grads."""
    import torch.nn.functional as F
    from mgproto_amd import ops
    g = torch.Generator().manual_seed(4)
    feat = F.normalize(torch.randn(64, 16, generator=g), dim=1)
    means = F.normalize(torch.rand(10, 16, generator=g), dim=1)
    covs = torch.full((10, 16), 0.4)

    f1 = feat.clone().requires_grad_(True)
    out1 = ops.gmm_scores(f1, means, covs, apply_exp=True)   # uniform path
    go = torch.randn_like(out1)
    out1.backward(go)

    os.environ['MGPROTO_NO_GMM_UNI'] = '1'
    try:
        f2 = feat.clone().requires_grad_(True)
        out2 = ops.gmm_scores(f2, means, covs, apply_exp=True)
        out2.backward(go)
    finally:
        del os.environ['MGPROTO_NO_GMM_UNI']
    assert torch.allclose(out1, out2, atol=1e-5, rtol=1e-5)
    assert torch.allclose(f1.grad, f2.grad, atol=1e-5, rtol=1e-5)


def test_gmm_nonuniform_covs_use_general_path():
    """Per-prototype sigma must NOT be treated as uniform."""
    import torch.nn.functional as F
    from mgproto_amd import ops
    from mgproto_amd.ops import reference as R
    g = torch.Generator().manual_seed(5)
    feat = F.normalize(torch.randn(32, 16, generator=g), dim=1)
    means = F.normalize(torch.rand(6, 16, generator=g), dim=1)
    covs = 0.3 + 0.5 * torch.rand(6, 16, generator=g)
    out = ops.gmm_scores(feat, means, covs, apply_exp=False)
    want = R.gmm_logprob(feat, means, covs)
    assert torch.allclose(out, want, atol=1e-5, rtol=1e-5)


# --------------------------------------- reducer across phase transitions

@pytest.mark.parametrize('world', [2, 4])
def test_reducer_warm_joint_fuzz(world):
    _run_workers(impl_reducer_warm_joint, world=world)


class _TwoPartNet(torch.nn.Module):
    """Stand-in for backbone(freezable)+head, sized to force multi-param
    buckets at bucket_mb=1."""

    def __init__(self):
        super().__init__()
        self.features = torch.nn.Sequential(
            torch.nn.Linear(64, 256), torch.nn.ReLU(),
            torch.nn.Linear(256, 256), torch.nn.ReLU(),
            torch.nn.Linear(256, 64))
        self.head = torch.nn.Sequential(
            torch.nn.Linear(64, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))

    def forward(self, x):
        return self.head(self.features(x))


def impl_reducer_warm_joint(rank, world):
    from mgproto_amd.parallel import Comm, BucketedGradReducer
    comm = Comm(backend='gloo')
    torch.manual_seed(0)
    net = _TwoPartNet()
    reducer = BucketedGradReducer(net, comm, bucket_mb=1)

    # oracle: a single-process copy fed the concatenation of all ranks'
    # batches (grad of the mean loss == mean of per-rank grads when every
    # rank contributes the same batch size — guaranteed by the sampler)
    torch.manual_seed(0)
    oracle = _TwoPartNet()

    phases = ['warm', 'warm', 'joint', 'warm', 'joint', 'joint']
    last = None
    for step, phase in enumerate(phases):
        for p in net.features.parameters():
            p.requires_grad = phase == 'joint'
        for p in oracle.features.parameters():
            p.requires_grad = phase == 'joint'
        if phase != last:
            reducer.rebuild()
        last = phase

        g = torch.Generator().manual_seed(1000 + step)
        xs = torch.randn(world * 8, 64, generator=g)
        x = xs[rank * 8:(rank + 1) * 8]

        reducer.prepare()
        for p in net.parameters():
            p.grad = None
        net(x).pow(2).mean().backward()
        reducer.finalize()

        for p in oracle.parameters():
            p.grad = None
        oracle(xs).pow(2).mean().backward()

        for (n, p), (no, po) in zip(net.named_parameters(),
                                    oracle.named_parameters()):
            assert n == no
            if p.grad is None:
                assert po.grad is None or phase == 'warm', n
                continue
            assert torch.allclose(p.grad, po.grad, atol=1e-6), \
                (phase, n, (p.grad - po.grad).abs().max().item())
    return True


# ------------------------------------------------ sampler property fuzz

def test_train_sampler_invariants_fuzz():
    """Random (N, world, epoch): every rank gets the same count, the union
    covers the dataset, and padding never exceeds world-1 duplicates."""
    import random
    rng = random.Random(0)
    for _ in range(40):
        n = rng.randint(1, 300)
        world = rng.choice([2, 3, 4, 5, 8])
        epoch = rng.randint(0, 10)
        per_rank = _sampler_epoch_indices(n, world, epoch)
        lens = {len(ix) for ix in per_rank}
        assert len(lens) == 1, (n, world, lens)
        allix = [i for ix in per_rank for i in ix]
        assert set(allix) == set(range(n)), (n, world)
        dup = len(allix) - n
        assert 0 <= dup < world, (n, world, dup)


def test_faithful_aug_reachable_from_settings(tmp_path):
    """ADVICE low: cfg.fast_augment=False must select the reference's
    4-pass chain in the real image loaders."""
    from PIL import Image
    d = tmp_path / 'class_0'
    d.mkdir()
    Image.new('RGB', (40, 40), (10, 20, 30)).save(d / 'a.jpg')
    from mgproto_amd.settings import Settings
    from mgproto_amd.data.loaders import build_image_loaders
    from mgproto_amd.data import transforms as T
    for fast, expect in ((True, T.FusedTrainTransform), (False, T.Compose)):
        cfg = Settings(img_size=32, num_workers=0, train_batch_size=2,
                       fast_augment=fast)
        cfg.train_dir = cfg.train_push_dir = cfg.test_dir = str(tmp_path)
        train_loader, *_ = build_image_loaders(cfg)
        assert isinstance(train_loader.dataset.transform, expect), fast
