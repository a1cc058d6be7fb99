"""world_size=8 CPU (gloo) rehearsal of the driver's 8-GPU SCALE ladder.

The round-end scaling bench runs bench.py at N=1/2/4/8 on one node; the
2- and 4-rank gloo tests cover the collective logic, but 8 ranks is the
shape that actually ships — these verify the reducer, the padded
sampler, the enqueue bit-identity and the push merge at that width
(8 spawned procs; tiny problem sizes keep it CPU-cheap)."""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _run_workers(fn, world=8, extra=()):
    ctx = mp.get_context('spawn')
    port = str(29960 + (os.getpid() + hash(fn.__name__)) % 30)
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
        p.join(timeout=120)
    return results


def _worker_entry(fn_name, rank, world, port, q, extra):
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world)
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = port
    torch.manual_seed(200 + rank)
    torch.set_num_threads(1)          # 8 procs on one box
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


def test_reducer_and_phases_world8():
    _run_workers(impl_reducer_phases)


def impl_reducer_phases(rank, world):
    from mgproto_amd.parallel import Comm, BucketedGradReducer
    comm = Comm(backend='gloo')
    torch.manual_seed(0)
    net = torch.nn.Sequential(torch.nn.Linear(24, 48), torch.nn.ReLU(),
                              torch.nn.Linear(48, 24), torch.nn.ReLU(),
                              torch.nn.Linear(24, 4))
    reducer = BucketedGradReducer(net, comm, bucket_mb=1)
    torch.manual_seed(0)
    oracle = torch.nn.Sequential(torch.nn.Linear(24, 48), torch.nn.ReLU(),
                                 torch.nn.Linear(48, 24), torch.nn.ReLU(),
                                 torch.nn.Linear(24, 4))
    for step, phase in enumerate(['warm', 'joint', 'warm', 'joint']):
        frozen = phase == 'warm'
        for p in net[0].parameters():
            p.requires_grad = not frozen
        for p in oracle[0].parameters():
            p.requires_grad = not frozen
        reducer.rebuild()

        g = torch.Generator().manual_seed(500 + step)
        xs = torch.randn(world * 4, 24, generator=g)
        x = xs[rank * 4:(rank + 1) * 4]
        reducer.prepare()
        for p in net.parameters():
            p.grad = None
        net(x).pow(2).mean().backward()
        reducer.finalize()
        for p in oracle.parameters():
            p.grad = None
        oracle(xs).pow(2).mean().backward()
        for p, po in zip(net.parameters(), oracle.parameters()):
            if p.grad is None:
                continue
            assert torch.allclose(p.grad, po.grad, atol=1e-6)
    return True


def test_enqueue_banks_identical_world8():
    _run_workers(impl_enqueue_banks)


def impl_enqueue_banks(rank, world):
    import torch.nn.functional as F
    from mgproto_amd.model import construct_MGProto
    from mgproto_amd.parallel import Comm, make_dp_correct
    comm = Comm(backend='gloo')
    torch.manual_seed(0)
    C = 4
    m = construct_MGProto('resnet18', pretrained=False, img_size=32,
                          prototype_shape=(8, 16, 1, 1), num_classes=C,
                          add_on_layers_type='regular', sz_embedding=8,
                          mem_capacity=8, mine_K=2)
    m = make_dp_correct(m, comm, train_batch_size=4)
    g = torch.Generator().manual_seed(900 + rank)
    for _ in range(3):
        x = torch.randn(4, 3, 32, 32, generator=g)
        y = torch.randint(0, C, (4,), generator=g)
        with torch.no_grad():
            m(x, y)
    # banks must be bit-identical on all 8 ranks
    digest = m.queue.mem.flatten().double().sum()
    all_d = comm.all_gather_fixed(digest.unsqueeze(0)).flatten()
    assert torch.allclose(all_d, all_d[0].expand(world)), all_d
    mem0 = comm.all_gather_fixed(m.queue.mem.unsqueeze(0))[0, 0]
    assert torch.equal(mem0, m.queue.mem)
    return True


def test_sampler_world8_equal_batches():
    """Padded sampler at 8 ranks: equal step counts for awkward N."""
    from mgproto_amd.data.loaders import make_train_sampler

    class _D(torch.utils.data.Dataset):
        def __init__(self, n):
            self.n = n

        def __len__(self):
            return self.n

        def __getitem__(self, i):
            return i

    for n in (17, 64, 100, 257):
        lens = set()
        union = set()
        for rank in range(8):
            s = make_train_sampler(_D(n), 8, rank)
            s.set_epoch(3)
            ix = list(iter(s))
            lens.add(len(ix))
            union.update(ix)
        assert len(lens) == 1, (n, lens)
        assert union == set(range(n)), n


def test_push_merge_world8():
    _run_workers(impl_push_merge)


def impl_push_merge(rank, world):
    """Candidate gather + greedy merge is rank-count invariant at 8."""
    import numpy as np
    from mgproto_amd.parallel import Comm
    from mgproto_amd.parallel.state_sync import gather_push_candidates
    comm = Comm(backend='gloo')
    # deterministic global candidate set, stride-sharded like push
    g = torch.Generator().manual_seed(77)
    P = 6
    dists_all = torch.rand(48, generator=g)
    meta_all = torch.stack([torch.randint(0, P, (48,), generator=g),
                            torch.arange(48),
                            torch.randint(0, 4, (48,), generator=g),
                            torch.randint(0, 4, (48,), generator=g)], dim=1)
    mine = slice(rank, 48, world)
    d_all, m_all = gather_push_candidates(comm, dists_all[mine],
                                          meta_all[mine])
    # merge exactly as engine/push.py does
    d_np, m_np = d_all.numpy(), m_all.numpy()
    order = np.lexsort((m_np[:, 3], m_np[:, 2], m_np[:, 1], d_np))
    per = {j: [] for j in range(P)}
    for oi in order:
        per[int(m_np[oi, 0])].append(oi)
    claimed, chosen = set(), []
    for j in range(P):
        for oi in per[j]:
            img = int(m_np[oi, 1])
            if img in claimed:
                continue
            claimed.add(img)
            chosen.append((j, img, int(m_np[oi, 2]), int(m_np[oi, 3])))
            break
    # every rank must agree, and the result must equal the single-process
    # merge over the unsharded arrays
    d1, m1 = dists_all.numpy(), meta_all.numpy()
    order1 = np.lexsort((m1[:, 3], m1[:, 2], m1[:, 1], d1))
    per1 = {j: [] for j in range(P)}
    for oi in order1:
        per1[int(m1[oi, 0])].append(oi)
    claimed1, chosen1 = set(), []
    for j in range(P):
        for oi in per1[j]:
            img = int(m1[oi, 1])
            if img in claimed1:
                continue
            claimed1.add(img)
            chosen1.append((j, img, int(m1[oi, 2]), int(m1[oi, 3])))
            break
    assert chosen == chosen1, (chosen, chosen1)
    return True
