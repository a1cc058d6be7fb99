"""DP-correct model-state synchronization (memory bank, EM, push).

The reference mutates model state inside ``forward`` (memory-bank enqueue,
model.py:228-252) and under DataParallel those replica writes are silently
lost. Here the mutation sites are made rank-invariant:

* **enqueue**: every rank's per-batch candidates are padded to a fixed
  shape, all-gathered (one fixed-size RCCL collective, no host sync), and
  pushed by EVERY rank in (rank, class, sample) order — all ranks hold
  bit-identical banks.
* **EM**: with identical banks, ``update_GMM`` is replicated deterministic
  compute (cheap, ~1 GFLOP) — zero communication, no drift.
* **push**: per-rank candidate lists are all-gathered and merged with a
  deterministic global greedy (engine/push.py), so prototype assignments
  are independent of GPU count (SURVEY.md hard part #4).
"""

from typing import Tuple

import torch

from .comm import Comm


class DistributedEnqueue:
    """Replaces ``model._enqueue_fn``: all-gather + replicated push.

    The payload is a fixed ``[max_items, d+1]`` buffer per rank (feature
    rows + label column); unused rows carry the sentinel label C, which
    MemoryBank.push discards device-side. One ``all_gather_into_tensor``
    per training step (~max_items*(d+1)*4 bytes per rank, e.g. 800*65*4
    ≈ 208 KB — negligible over xGMI).
    """

    def __init__(self, model, comm: Comm, max_items: int):
        self.model = model
        self.comm = comm
        self.max_items = max_items
        self.C = model.num_classes
        d = model.prototype_shape[1]
        self.buf = torch.zeros(max_items, d + 1, device=comm.device)

    @torch.no_grad()
    def __call__(self, feats: torch.Tensor, labels: torch.Tensor):
        M = feats.shape[0]
        assert M <= self.max_items, (M, self.max_items)
        self.buf[:, -1] = float(self.C)          # sentinel-fill labels
        self.buf[:M, :-1] = feats
        self.buf[:M, -1] = labels.to(self.buf.dtype)
        gathered = self.comm.all_gather_fixed(self.buf)      # [W, max, d+1]
        flat = gathered.reshape(-1, self.buf.shape[1])
        all_feats = flat[:, :-1]
        all_labels = flat[:, -1].to(torch.int64)
        self.model.queue.push(all_feats, all_labels)
        self.model._memory_updated.index_fill_(0, all_labels, True)


def make_dp_correct(model, comm: Comm, train_batch_size: int):
    """Install DP-correct enqueue on the model; broadcast initial state."""
    if not comm.is_distributed:
        return model
    comm.broadcast_module(model)
    max_items = train_batch_size * model.num_prototypes_per_class
    model._enqueue_fn = DistributedEnqueue(model, comm, max_items)
    return model


def gather_push_candidates(comm: Comm, dists: torch.Tensor,
                           meta: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """All-gather per-rank push candidate arrays (varlen along dim 0).

    ``dists``: [M] fp32 candidate distances; ``meta``: [M, 4] int64
    (prototype j, global image idx, h, w). Returns concatenated arrays in
    rank order.
    """
    if not comm.is_distributed:
        return dists, meta
    # collectives must run on the backend's device: push.py collects the
    # candidates on the host, but an RCCL process group rejects CPU
    # tensors — stage through comm.device and hand CPU results back to
    # the (host-side) merge
    dev = comm.device
    flat_meta = meta.reshape(-1).to(device=dev, dtype=torch.float64)
    d_all = comm.all_gather_varlen(dists.to(dev).double())
    m_all = comm.all_gather_varlen(flat_meta)
    return d_all.float().cpu(), m_all.view(-1, 4).to(torch.int64).cpu()
