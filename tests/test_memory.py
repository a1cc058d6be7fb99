"""MemoryBank semantics vs a literal FIFO oracle (reference utils/memory.py)."""

import torch

from mgproto_amd.utils.memory import MemoryBank


class FifoOracle:
    """Literal per-class FIFO with capacity cap (reference memory.py:32-73)."""

    def __init__(self, C, cap):
        self.items = {c: [] for c in range(C)}
        self.cap = cap

    def push(self, feats, labels):
        for f, c in zip(feats, labels.tolist()):
            self.items[c].append(f.clone())
            if len(self.items[c]) > self.cap:
                self.items[c].pop(0)

    def logical(self, c):
        if not self.items[c]:
            return torch.empty(0)
        return torch.stack(self.items[c])


def test_push_pull_fifo():
    C, cap, d = 5, 8, 4
    bank = MemoryBank(C, d, capacity=C * cap)
    oracle = FifoOracle(C, cap)
    g = torch.Generator().manual_seed(0)
    for _ in range(20):
        M = int(torch.randint(1, 15, (1,), generator=g))
        feats = torch.randn(M, d, generator=g)
        labels = torch.randint(0, C, (M,), generator=g)
        bank.push(feats, labels)
        oracle.push(feats, labels)
    for c in range(C):
        want = oracle.logical(c)
        assert int(bank.mem_len[c]) == want.shape[0]
        if want.numel():
            got = bank._logical(c)
            assert torch.allclose(got, want), f'class {c} mismatch'


def test_pull_all_labels():
    C, cap, d = 3, 4, 2
    bank = MemoryBank(C, d, capacity=C * cap)
    bank.push(torch.randn(5, d), torch.tensor([0, 0, 2, 2, 2]))
    data, labels = bank.pull_all()
    assert data.shape == (5, d)
    assert labels.tolist() == [0, 0, 2, 2, 2]


def test_oversized_single_push_keeps_newest():
    C, cap, d = 2, 3, 2
    bank = MemoryBank(C, d, capacity=C * cap)
    feats = torch.arange(10, dtype=torch.float32).view(5, 2)
    bank.push(feats, torch.zeros(5, dtype=torch.long))
    # newest cap items survive (deterministic divergence from the
    # reference's random subsample, memory.py:52)
    got = bank._logical(0)
    assert torch.equal(got, feats[2:])


def test_state_dict_reference_layout():
    C, cap, d = 3, 4, 2
    bank = MemoryBank(C, d, capacity=C * cap)
    bank.push(torch.randn(6, d), torch.tensor([0, 0, 1, 1, 1, 2]))
    sd = bank.state_dict()
    assert set(sd.keys()) == {'cls0', 'cls1', 'cls2', 'mem_len'}
    assert sd['cls0'].shape == (cap, d)
    # roundtrip into a fresh bank
    bank2 = MemoryBank(C, d, capacity=C * cap)
    bank2.load_state_dict(sd)
    for c in range(C):
        assert torch.allclose(bank2._logical(c), bank._logical(c))
    # FIFO continues correctly after a load
    bank.push(torch.randn(3, d), torch.tensor([1, 1, 1]))
    bank2.push(bank.mem.new_zeros(0, d), torch.zeros(0, dtype=torch.long))
    assert int(bank2.mem_len[1]) == 3


def test_wraparound_after_full():
    C, cap, d = 1, 3, 1
    bank = MemoryBank(C, d, capacity=cap)
    oracle = FifoOracle(C, cap)
    for i in range(7):
        f = torch.tensor([[float(i)]])
        l = torch.tensor([0])
        bank.push(f, l)
        oracle.push(f, l)
    assert torch.allclose(bank._logical(0), oracle.logical(0))


def test_load_capacity_mismatch_reports_cleanly():
    """A wrong-capacity checkpoint surfaces through load_state_dict's
    error list (with a pointer to the fix), not a mid-hook crash."""
    import pytest

    a = MemoryBank(2, 4, capacity=8)    # cap 4/class
    b = MemoryBank(2, 4, capacity=16)   # cap 8/class
    sd = a.state_dict()
    with pytest.raises(RuntimeError, match='mem_capacity'):
        b.load_state_dict(sd)
