"""Multi-process DP correctness on CPU (gloo, world_size=2).

Covers SURVEY.md §2.2 C1-C4: bucketed gradient all-reduce, DP-correct
memory-bank enqueue (identical banks on all ranks), replicated EM
determinism, distributed push candidate merge, and metric reduction —
the paths the driver exercises on 8 GPUs with RCCL.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _run_workers(fn, world=2, extra=()):
    ctx = mp.get_context('spawn')
    port = str(29600 + (os.getpid() + hash(fn.__name__)) % 200)
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
    except Exception as e:  # noqa: BLE001
        import traceback
        q.put((rank, 'ERROR ' + traceback.format_exc()))
        raise
    finally:
        import torch.distributed as dist
        if dist.is_initialized():
            dist.destroy_process_group()


# ---------------------------------------------------------------- reducer

def test_grad_reducer_matches_mean():
    _run_workers(impl_grad_reducer_matches_mean)


def impl_grad_reducer_matches_mean(rank, world):
    from mgproto_amd.parallel import Comm, BucketedGradReducer
    comm = Comm(backend='gloo')
    torch.manual_seed(0)  # same init everywhere
    net = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.ReLU(),
                              torch.nn.Linear(32, 4))
    reducer = BucketedGradReducer(net, comm, bucket_mb=1)

    torch.manual_seed(1000 + rank)  # different data per rank
    x = torch.randn(8, 16)
    y = torch.randn(8, 4)

    reducer.prepare()
    loss = ((net(x) - y) ** 2).mean()
    loss.backward()
    reducer.finalize()
    got = [p.grad.clone() for p in net.parameters()]

    # oracle: gather per-rank grads, average
    net.zero_grad(set_to_none=True)
    loss = ((net(x) - y) ** 2).mean()
    loss.backward()
    local = [p.grad.clone() for p in net.parameters()]
    import torch.distributed as dist
    want = []
    for g in local:
        t = g.clone()
        dist.all_reduce(t)
        want.append(t / world)
    for g1, g2 in zip(got, want):
        assert torch.allclose(g1, g2, atol=1e-6)
    return 'ok'


def test_grad_reducer_partial_bucket():
    _run_workers(impl_grad_reducer_partial_bucket)


def impl_grad_reducer_partial_bucket(rank, world):
    """Frozen params mid-bucket must not hang or corrupt reduction."""
    from mgproto_amd.parallel import Comm, BucketedGradReducer
    comm = Comm(backend='gloo')
    torch.manual_seed(0)
    net = torch.nn.Sequential(torch.nn.Linear(8, 8), torch.nn.Linear(8, 8))
    reducer = BucketedGradReducer(net, comm, bucket_mb=1024)  # one bucket
    # freeze the second layer AFTER reducer construction (warm phase)
    for p in net[1].parameters():
        p.requires_grad = False
    torch.manual_seed(2000 + rank)
    x = torch.randn(4, 8)
    reducer.prepare()
    net(x).sum().backward()
    reducer.finalize()
    g = net[0].weight.grad.clone()
    import torch.distributed as dist
    want = net[0].weight.grad.clone()
    dist.all_reduce(want)
    want /= world
    assert torch.allclose(g, want, atol=1e-6)
    return 'ok'


# ----------------------------------------------------------- enqueue + EM

def _tiny_model():
    from mgproto_amd.model import construct_MGProto
    torch.manual_seed(0)
    return construct_MGProto('resnet18', pretrained=False, img_size=64,
                             prototype_shape=(20, 16, 1, 1), num_classes=5,
                             add_on_layers_type='regular', sz_embedding=8,
                             mem_capacity=6, mine_K=3)


def test_distributed_enqueue_identical_banks():
    _run_workers(impl_distributed_enqueue_identical_banks)


def impl_distributed_enqueue_identical_banks(rank, world):
    from mgproto_amd.parallel import Comm, make_dp_correct
    comm = Comm(backend='gloo')
    model = _tiny_model()
    model = make_dp_correct(model, comm, train_batch_size=4)

    torch.manual_seed(3000 + rank)
    x = torch.randn(4, 3, 64, 64)
    gt = torch.randint(0, 5, (4,))
    with torch.no_grad():
        model(x, gt)

    # all ranks must hold bit-identical banks and dirty flags
    import torch.distributed as dist
    state = torch.cat([model.queue.mem.flatten(),
                       model.queue.mem_len.float(),
                       model.queue.head.float(),
                       model.memory_updated_cls.float()])
    ref = state.clone()
    dist.broadcast(ref, src=0)
    assert torch.equal(state, ref), 'bank state diverged across ranks'

    # EM is replicated deterministic compute -> stays identical
    model.queue.mem_len.fill_(model.queue.cap_cls)  # force full
    model.update_GMM()
    means = model.prototype_means.data.clone()
    ref = means.clone()
    dist.broadcast(ref, src=0)
    assert torch.allclose(means, ref, atol=0), 'EM diverged across ranks'
    return 'ok'


def test_distributed_enqueue_matches_bigbatch():
    _run_workers(impl_distributed_enqueue_matches_bigbatch)


def impl_distributed_enqueue_matches_bigbatch(rank, world):
    """2-rank enqueue of per-rank batches == single push of the rank-ordered
    concatenation (the rank-invariance contract)."""
    from mgproto_amd.parallel import Comm, DistributedEnqueue
    from mgproto_amd.utils.memory import MemoryBank
    comm = Comm(backend='gloo')
    model = _tiny_model()
    enq = DistributedEnqueue(model, comm, max_items=12)

    torch.manual_seed(4000 + rank)
    feats = torch.randn(7, 16)
    labels = torch.randint(0, 5, (7,))
    enq(feats, labels)

    # oracle on every rank: gather both ranks' items, push in rank order
    import torch.distributed as dist
    all_f = [torch.zeros_like(feats) for _ in range(world)]
    all_l = [torch.zeros_like(labels) for _ in range(world)]
    dist.all_gather(all_f, feats)
    dist.all_gather(all_l, labels)
    oracle = MemoryBank(5, 16, capacity=5 * 6)
    oracle.push(torch.cat(all_f), torch.cat(all_l))
    for c in range(5):
        assert torch.allclose(model.queue._logical(c), oracle._logical(c))
    return 'ok'


# ------------------------------------------------------------------- push

def test_distributed_push_matches_single():
    _run_workers(impl_distributed_push_matches_single)


def impl_distributed_push_matches_single(rank, world):
    from mgproto_amd.parallel import Comm
    from mgproto_amd.engine import push_prototypes
    from mgproto_amd.data import SyntheticImages
    from torch.utils.data import DataLoader, Subset

    comm = Comm(backend='gloo')
    model = _tiny_model()
    model.eval()
    ds = SyntheticImages(n=8, num_classes=5, img_size=64, normalize=False)

    def collate(batch):
        return (torch.stack([b[0] for b in batch]),
                torch.tensor([b[1] for b in batch]),
                torch.tensor([b[2] for b in batch]))

    # rank shard: interleaved indices (DistributedSampler-style)
    shard = Subset(ds, list(range(rank, len(ds), world)))

    def collate_shard(batch):
        return collate(batch)

    loader = DataLoader(shard, batch_size=4, collate_fn=collate_shard)
    # Subset items keep their global idx in position 2 -> global identity ok
    chosen = push_prototypes(loader, model, log=lambda *a: None, comm=comm)

    # oracle: single-process push over the full dataset
    model2 = _tiny_model()
    model2.eval()
    full_loader = DataLoader(ds, batch_size=4, collate_fn=collate)
    chosen_single = push_prototypes(full_loader, model2, log=lambda *a: None)

    assert chosen == chosen_single, (chosen, chosen_single)
    assert torch.allclose(model.prototype_means.data,
                          model2.prototype_means.data, atol=1e-5)
    return 'ok'


# ---------------------------------------------------------------- metrics

def test_metric_allreduce():
    _run_workers(impl_metric_allreduce)


def impl_metric_allreduce(rank, world):
    from mgproto_amd.parallel import Comm
    comm = Comm(backend='gloo')
    t = torch.tensor([float(rank + 1), 10.0])
    out = comm.all_reduce_sum(t.clone())
    assert torch.allclose(out, torch.tensor([3.0, 20.0]))
    v = comm.all_gather_varlen(torch.arange(rank + 1).float())
    assert v.tolist() == [0.0, 0.0, 1.0]
    return 'ok'


def test_distributed_push_three_ranks():
    _run_workers(impl_distributed_push_matches_single, world=3)


def test_grad_reducer_four_ranks():
    _run_workers(impl_grad_reducer_matches_mean, world=4)


def test_engine_epoch_ranks_stay_identical():
    """Full engine epoch (tnt.train with reducer + DistributedEnqueue + EM)
    on 2 ranks with DIFFERENT data shards: all model/aux parameters and the
    memory bank must end bit-identical across ranks. Regression test for
    the reducer not being armed per step inside the engine loop."""
    res = _run_workers(impl_engine_epoch_ranks_stay_identical, world=2)
    for k in res[0]['digests']:
        assert res[0]['digests'][k] == res[1]['digests'][k], \
            f'rank drift in {k}'
    # different shards must actually move the weights (reduction is not a
    # no-op freeze)
    assert res[0]['moved'] > 0


def impl_engine_epoch_ranks_stay_identical(rank, world):
    from torch.utils.data import DataLoader

    from mgproto_amd.model import construct_MGProto
    from mgproto_amd.data import SyntheticImages
    from mgproto_amd.engine import joint
    from mgproto_amd.engine import train as tnt_train
    from mgproto_amd.losses import build_aux_loss
    from mgproto_amd.parallel import (Comm, BucketedGradReducer,
                                      make_dp_correct)

    comm = Comm(backend='gloo')
    C, K, d = 6, 2, 16
    torch.manual_seed(100 + rank)   # different init per rank on purpose:
    model = construct_MGProto('resnet34', pretrained=False, img_size=64,
                              prototype_shape=(C * K, d, 1, 1),
                              num_classes=C, add_on_layers_type='regular',
                              sz_embedding=8, mem_capacity=8, mine_K=2)
    aux = build_aux_loss('Proxy_Anchor', nb_classes=C, sz_embed=8)
    # ... make_dp_correct broadcasts rank 0's state
    model = make_dp_correct(model, comm, train_batch_size=4)
    comm.broadcast_module(aux)
    reducer = BucketedGradReducer([model, aux], comm, bucket_mb=1)
    opt = torch.optim.Adam([
        {'params': model.parameters(), 'lr': 1e-3},
        {'params': aux.parameters(), 'lr': 1e-3},
    ])
    joint(model)
    before = model.add_on_layers[0].weight.detach().clone()

    def _collate(batch):
        return (torch.stack([b[0] for b in batch]),
                torch.tensor([b[1] for b in batch]),
                torch.tensor([b[2] for b in batch]))

    ds = SyntheticImages(n=16, num_classes=C, img_size=64, seed=500 + rank)
    loader = DataLoader(ds, batch_size=4, collate_fn=_collate)
    # prefill the bank so the EM path runs this epoch (rank-invariant, as
    # DistributedEnqueue guarantees for the in-training enqueue path)
    g = torch.Generator().manual_seed(4242)
    for c in range(C):
        model.queue.push(torch.nn.functional.normalize(
            torch.randn(8, d, generator=g), dim=1),
            torch.full((8,), c, dtype=torch.long))
        model.memory_updated_cls[c] = True

    tnt_train(model, loader, opt, aux_criterion=aux, use_mine=True,
              update_GMM=True, coefs={'crs_ent': 1, 'mine': 0.2, 'aux': 0.5},
              log=lambda *a: None, amp_dtype='off', print_every=0,
              comm=comm, reducer=reducer)

    import hashlib

    def digest(t):
        return hashlib.sha256(
            t.detach().cpu().contiguous().float().numpy().tobytes()).hexdigest()

    digests = {f'model/{k}': digest(v) for k, v in model.state_dict().items()}
    digests.update({f'aux/{k}': digest(v)
                    for k, v in aux.state_dict().items()})
    moved = float((model.add_on_layers[0].weight.detach()
                   - before).abs().sum())
    return {'digests': digests, 'moved': moved}


def test_ood_eval_two_ranks_matches_single():
    """_testing_with_OoD on 2 rank-sharded loaders == single-process run
    over the full set (quantile threshold and FPR are order-invariant)."""
    res = _run_workers(impl_ood_eval_two_ranks, world=2)
    assert res[0] == res[1]                      # rank-invariant outputs
    single = impl_ood_eval_single()
    assert abs(res[0]['acc'] - single['acc']) < 1e-6
    assert abs(res[0]['FPR95_1'] - single['FPR95_1']) < 1e-6


def _ood_model_and_data():
    from torch.utils.data import DataLoader

    from mgproto_amd.data import SyntheticImages
    from mgproto_amd.model import construct_MGProto

    torch.manual_seed(11)
    C, K, d = 4, 2, 16
    model = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(C * K, d, 1, 1),
                              num_classes=C, add_on_layers_type='regular',
                              sz_embedding=8, mem_capacity=8, mine_K=2)
    model.eval()

    def collate(batch):
        return (torch.stack([b[0] for b in batch]),
                torch.tensor([b[1] for b in batch]),
                torch.tensor([b[2] for b in batch]))

    id_ds = SyntheticImages(n=16, num_classes=C, img_size=64, seed=5)
    ood_ds = SyntheticImages(n=8, num_classes=C, img_size=64, seed=9)
    return model, id_ds, ood_ds, collate, DataLoader


def impl_ood_eval_two_ranks(rank, world):
    from torch.utils.data import Subset

    from mgproto_amd.engine import _testing_with_OoD
    from mgproto_amd.parallel import Comm

    comm = Comm(backend='gloo')
    model, id_ds, ood_ds, collate, DataLoader = _ood_model_and_data()
    comm.broadcast_module(model)

    def shard(ds):
        return Subset(ds, list(range(rank, len(ds), world)))

    id_loader = DataLoader(shard(id_ds), batch_size=4, collate_fn=collate)
    ood_loader = DataLoader(shard(ood_ds), batch_size=4, collate_fn=collate)
    _, results = _testing_with_OoD(model, (id_loader, ood_loader),
                                   log=lambda *a: None, amp_dtype='off',
                                   comm=comm)
    return {'acc': float(results['acc']),
            'FPR95_1': float(results['FPR95_1'])}


def impl_ood_eval_single():
    from mgproto_amd.engine import _testing_with_OoD

    model, id_ds, ood_ds, collate, DataLoader = _ood_model_and_data()
    id_loader = DataLoader(id_ds, batch_size=4, collate_fn=collate)
    ood_loader = DataLoader(ood_ds, batch_size=4, collate_fn=collate)
    _, results = _testing_with_OoD(model, (id_loader, ood_loader),
                                   log=lambda *a: None, amp_dtype='off')
    return {'acc': float(results['acc']),
            'FPR95_1': float(results['FPR95_1'])}


def test_comm_gather_primitives():
    res = _run_workers(impl_comm_gather_primitives, world=2)
    assert res[0] == res[1] == 'ok'


def impl_comm_gather_primitives(rank, world):
    from mgproto_amd.parallel import Comm

    comm = Comm(backend='gloo')
    # fixed: [world, *shape], rank-major
    t = torch.full((2, 3), float(rank))
    g = comm.all_gather_fixed(t)
    assert g.shape == (2, 2, 3)
    assert torch.all(g[0] == 0.0) and torch.all(g[1] == 1.0)
    # scalar input gets a leading dim
    s = comm.all_gather_fixed(torch.tensor(float(rank)))
    assert s.flatten().tolist() == [0.0, 1.0]
    # varlen concatenates in rank order
    v = comm.all_gather_varlen(torch.arange(2 + rank).float())
    assert v.tolist() == [0.0, 1.0, 0.0, 1.0, 2.0]
    # async reduce completes
    x = torch.ones(4)
    w = comm.all_reduce_sum_async(x)
    w.wait()
    assert torch.all(x == world)
    # broadcast_buffers: rank 0's float buffers win; ints untouched
    m = torch.nn.BatchNorm2d(3)
    m.running_mean.fill_(float(rank))
    m.num_batches_tracked.fill_(rank)
    comm.broadcast_buffers(m)
    assert torch.all(m.running_mean == 0.0)
    assert int(m.num_batches_tracked) == rank   # non-float: left alone
    return 'ok'


def test_grad_reducer_random_configs():
    """Reducer over randomized layer shapes and bucket sizes (1 bucket ..
    one-per-param) against manually averaged gradients, several steps."""
    res = _run_workers(impl_grad_reducer_random_configs, world=2)
    assert res[0] == res[1] == 'ok'


def impl_grad_reducer_random_configs(rank, world):
    import torch.distributed as dist

    from mgproto_amd.parallel import Comm, BucketedGradReducer

    comm = Comm(backend='gloo')
    for trial in range(4):
        g = torch.Generator().manual_seed(900 + trial)   # same on all ranks
        sizes = [int(torch.randint(1, 2000, (1,), generator=g)) for _ in range(6)]
        layers = []
        d_in = 8
        for s in sizes:
            layers.append(torch.nn.Linear(d_in, s))
            d_in = s
        model = torch.nn.Sequential(*layers)
        comm.broadcast_module(model)
        bucket_mb = [1, 1000][trial % 2]   # many tiny buckets vs one big
        red = BucketedGradReducer(model, comm, bucket_mb=bucket_mb)
        for step in range(2):
            x = torch.randn(3, 8, generator=torch.Generator().manual_seed(
                rank * 17 + step + trial))        # different data per rank
            red.prepare()
            model.zero_grad(set_to_none=True)
            model(x).square().mean().backward()
            # manual expectation BEFORE finalize touches grads
            expected = [p.grad.clone() for p in model.parameters()]
            for e in expected:
                dist.all_reduce(e)
                e.div_(world)
            red.finalize()
            for p, e in zip(model.parameters(), expected):
                assert torch.allclose(p.grad, e, atol=1e-6), \
                    (trial, step, bucket_mb)
        red.remove()
    return 'ok'


def test_distributed_enqueue_random_streams():
    """Randomized multi-step enqueue streams on 2 ranks: banks stay
    bit-identical AND equal a single-process push of the rank-ordered
    concatenation of every step's candidates."""
    res = _run_workers(impl_distributed_enqueue_random_streams, world=2)
    assert res[0] == res[1] == 'ok'


def impl_distributed_enqueue_random_streams(rank, world):
    from mgproto_amd.model import construct_MGProto
    from mgproto_amd.parallel import Comm
    from mgproto_amd.parallel.state_sync import DistributedEnqueue
    from mgproto_amd.utils.memory import MemoryBank

    comm = Comm(backend='gloo')
    C, K, d, cap = 5, 2, 8, 4
    torch.manual_seed(0)
    model = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(C * K, d, 1, 1),
                              num_classes=C, add_on_layers_type='regular',
                              sz_embedding=8, mem_capacity=cap, mine_K=2)
    enq = DistributedEnqueue(model, comm, max_items=6)
    shadow = MemoryBank(C, d, capacity=C * cap)   # single-process oracle

    for step in range(12):
        per_rank = []
        for r in range(world):
            g = torch.Generator().manual_seed(step * 100 + r)
            M = int(torch.randint(0, 7, (1,), generator=g))
            feats = torch.randn(M, d, generator=g)
            labels = torch.randint(0, C, (M,), generator=g)
            per_rank.append((feats, labels))
        feats, labels = per_rank[rank]
        enq(feats, labels)
        # oracle: rank-ordered concatenation with sentinel padding, exactly
        # as the fixed-size all-gather delivers it
        pf, pl = [], []
        for f, l in per_rank:
            pad = 6 - f.shape[0]
            pf.append(torch.cat([f, torch.zeros(pad, d)]))
            pl.append(torch.cat([l, torch.full((pad,), C, dtype=torch.long)]))
        shadow.push(torch.cat(pf), torch.cat(pl))

    assert torch.equal(model.queue.mem, shadow.mem)
    assert torch.equal(model.queue.mem_len, shadow.mem_len)
    # cross-rank bit-identity
    other = comm.all_gather_fixed(model.queue.mem.reshape(1, -1))
    assert torch.equal(other[0], other[1])
    return 'ok'


def test_distributed_push_empty_rank_shard():
    """More ranks than images: a rank with an EMPTY push shard must not
    deadlock or diverge — its varlen gathers contribute zero candidates."""
    _run_workers(impl_distributed_push_empty_shard, world=3)


def impl_distributed_push_empty_shard(rank, world):
    from torch.utils.data import DataLoader, Subset

    from mgproto_amd.data import SyntheticImages
    from mgproto_amd.engine import push_prototypes
    from mgproto_amd.parallel import Comm

    comm = Comm(backend='gloo')
    model = _tiny_model()
    model.eval()
    ds = SyntheticImages(n=2, num_classes=5, img_size=64, normalize=False)

    def collate(batch):
        return (torch.stack([b[0] for b in batch]),
                torch.tensor([b[1] for b in batch]),
                torch.tensor([b[2] for b in batch]))

    shard = Subset(ds, list(range(rank, len(ds), world)))   # rank 2: empty
    loader = DataLoader(shard, batch_size=2, collate_fn=collate)
    chosen = push_prototypes(loader, model, log=lambda *a: None, comm=comm)

    model2 = _tiny_model()
    model2.eval()
    chosen_single = push_prototypes(DataLoader(ds, batch_size=2,
                                               collate_fn=collate),
                                    model2, log=lambda *a: None)
    assert chosen == chosen_single
    assert torch.allclose(model.prototype_means.data,
                          model2.prototype_means.data, atol=1e-5)
    return 'ok'
