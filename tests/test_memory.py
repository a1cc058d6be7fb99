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


def test_pull_fix_class_and_dense():
    bank = MemoryBank(3, 4, capacity=6, mode='all')   # cap 2/class
    for c in (0, 2):
        bank.push(torch.full((2, 4), float(c + 1)),
                  torch.full((2,), c, dtype=torch.long))
    # fix_class: only flagged classes, logical order
    data, labels = bank.pull_fix_class(torch.tensor([1, 0, 1]))
    assert labels.tolist() == [0, 0, 2, 2]
    assert torch.all(data[:2] == 1.0) and torch.all(data[2:] == 3.0)
    # empty flag set
    d2, l2 = bank.pull_fix_class(torch.tensor([0, 1, 0]))
    assert d2 is None and l2 is None
    # pull_dense requires full classes; physical order is fine for EM
    dense = bank.pull_dense(torch.tensor([0, 2]))
    assert dense.shape == (2, 2, 4)
    assert torch.all(dense[0] == 1.0) and torch.all(dense[1] == 3.0)
    assert bank.full_mask().tolist() == [True, False, True]


def test_pull_fix_length_shapes():
    torch.manual_seed(0)
    bank = MemoryBank(3, 4, capacity=6, fix_length_mult=2)  # pull_num = 4
    for c in range(3):
        bank.push(torch.randn(2, 4), torch.full((2,), c, dtype=torch.long))
    onehot = torch.tensor([[1., 0., 1.], [0., 1., 0.]])
    out, lab = bank.pull_fix_length(onehot)
    assert lab is None
    assert out.shape == (4, 2, 4)    # [pull_num, B, d]
    # a row flagging a class with no data -> None
    bank2 = MemoryBank(3, 4, capacity=6)
    bank2.push(torch.randn(2, 4), torch.zeros(2, dtype=torch.long))
    assert bank2.pull_fix_length(torch.tensor([[0., 1., 0.]]))[0] is None
