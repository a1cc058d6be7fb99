"""Property-based fuzzing (hypothesis) of the stateful/batched ops whose
vectorized rewrites are easiest to get subtly wrong: the memory-bank ring,
the dedup enqueue, and the reference topk layout."""

import numpy as np
import pytest
import torch

hypothesis = pytest.importorskip('hypothesis')
from hypothesis import given, settings, strategies as st

from mgproto_amd.utils.memory import MemoryBank
from mgproto_amd.ops import reference as R


class FifoOracle:
    def __init__(self, C, cap):
        self.items = {c: [] for c in range(C)}
        self.cap = cap

    def push(self, feats, labels):
        for f, c in zip(feats, labels.tolist()):
            if c >= len(self.items):      # sentinel: discarded
                continue
            self.items[c].append(float(f[0]))
            if len(self.items[c]) > self.cap:
                self.items[c].pop(0)


@settings(max_examples=40, deadline=None)
@given(st.lists(st.lists(st.integers(min_value=0, max_value=4),  # 4 = sentinel
                         min_size=1, max_size=12),
                min_size=1, max_size=12))
def test_memory_bank_fuzz(pushes):
    """Arbitrary push sequences incl. sentinel labels == python FIFO."""
    C, cap = 4, 3
    bank = MemoryBank(C, 1, capacity=C * cap)
    oracle = FifoOracle(C, cap)
    counter = [0.0]
    for labels in pushes:
        feats = torch.tensor([[counter[0] + i] for i in range(len(labels))],
                             dtype=torch.float32)
        counter[0] += 100
        lab = torch.tensor(labels, dtype=torch.int64)
        bank.push(feats, lab)
        oracle.push(feats, lab)
    for c in range(C):
        got = bank._logical(c).flatten().tolist()
        assert got == oracle.items[c], (c, got, oracle.items[c])


@settings(max_examples=30, deadline=None)
@given(st.integers(min_value=1, max_value=7),   # B
       st.integers(min_value=1, max_value=4),   # K
       st.integers(min_value=2, max_value=9),   # HW
       st.integers(min_value=0, max_value=10_000))
def test_enqueue_candidates_fuzz(B, K, HW, seed):
    """Fixed-size enqueue == reference per-sample python-loop semantics."""
    C = 3
    g = torch.Generator().manual_seed(seed)
    feat = torch.randn(B * HW, 4, generator=g)
    top1 = torch.randint(0, HW, (B, C * K), generator=g)
    gt = torch.randint(0, C, (B,), generator=g)
    feats, labs = R.enqueue_candidates(feat, top1, gt, C, K, HW)
    assert labs.shape == (B * K,)

    # oracle (reference model.py:230-249): class asc, batch order, unique asc
    want_rows, want_labs = [], []
    for c in sorted(gt.unique().tolist()):
        for b in range(B):
            if gt[b] != c:
                continue
            for v in torch.unique(top1[b, c * K:(c + 1) * K]):
                want_rows.append(feat[b * HW + v])
                want_labs.append(c)
    real = labs < C
    assert labs[real].tolist() == want_labs
    assert torch.allclose(feats[real], torch.stack(want_rows))


@settings(max_examples=30, deadline=None)
@given(st.integers(min_value=1, max_value=4),
       st.integers(min_value=2, max_value=30),
       st.integers(min_value=1, max_value=8),
       st.integers(min_value=0, max_value=10_000))
def test_topk_reference_fuzz(B, HW, P, seed):
    g = torch.Generator().manual_seed(seed)
    T = min(HW, 5)
    probs = torch.rand(B, HW, P, generator=g)
    vals, idx = R.topk_hw(probs, T)
    # per (b, p): sorted desc, matches numpy partial sort
    arr = probs.numpy()
    for b in range(B):
        for p in range(P):
            col = arr[b, :, p]
            want = np.sort(col)[::-1][:T]
            assert np.allclose(vals[b, p].numpy(), want)
            assert np.allclose(col[idx[b, p].numpy()], want)


@settings(max_examples=25, deadline=None)
@given(st.integers(min_value=1, max_value=3),
       st.integers(min_value=2, max_value=6),
       st.integers(min_value=0, max_value=10_000))
def test_em_grads_match_autograd_fuzz(G, K, seed):
    g = torch.Generator().manual_seed(seed)
    N, d = 12, 6
    x = torch.nn.functional.normalize(torch.randn(G, N, d, generator=g), dim=2)
    means = torch.nn.functional.normalize(torch.rand(G, K, d, generator=g), dim=2)
    covs = 0.3 + 0.5 * torch.rand(G, K, d, generator=g)
    pi = torch.softmax(torch.rand(G, K, generator=g), dim=1)
    wlp, log_resp = R.em_e_step(x, means, covs, pi)
    grad, _ = R.em_m_step_grads(x, log_resp, wlp, means, covs)

    import math
    for gi in range(G):
        mu = means[gi].clone().requires_grad_(True)
        resp = torch.exp(log_resp[gi])
        resp = (resp + 0.1) / (resp + 0.1).sum(1, keepdim=True)
        diff = (x[gi].unsqueeze(1) - mu) / (covs[gi] + 1e-10)
        ll = (-0.5 * d * math.log(2 * math.pi)
              - torch.log(covs[gi] + 1e-10).sum(-1)
              - 0.5 * diff.pow(2).sum(-1)) + torch.log(pi[gi] + 1e-10)
        wll = -(resp.detach() * ll).sum(1).mean(0)
        pd = torch.cdist(mu, mu) ** 2
        I = 1 - torch.eye(K)
        div = (torch.exp(-pd) * I).sum() / I.sum()
        (wll + div).backward()
        assert torch.allclose(grad[gi], mu.grad, atol=1e-4, rtol=1e-3), \
            (grad[gi] - mu.grad).abs().max()


@settings(max_examples=20, deadline=None)
@given(st.integers(min_value=16, max_value=96),
       st.integers(min_value=16, max_value=96),
       st.integers(min_value=0, max_value=10_000))
def test_fastaug_native_vs_python_fuzz(H, W, seed):
    """Native C++ warp core vs the PIL fallback across random image sizes
    and random geometric draws (jitter off -> only the uint8 round-trip
    separates them)."""
    import os
    import random

    import numpy as np
    from PIL import Image

    from mgproto_amd.data import transforms as T

    if T._fastaug() is None:
        import pytest
        pytest.skip('native fastaug not built')
    rng = np.random.default_rng(seed)
    img = Image.fromarray(rng.integers(0, 255, (H, W, 3), dtype=np.uint8))
    tf = T.FusedTrainTransform(32, jitter=None)
    random.seed(seed)
    a = tf(img)
    os.environ['MGPROTO_NO_FASTAUG'] = '1'
    T._FASTAUG[:] = [None, False]
    try:
        random.seed(seed)
        b = tf(img)
    finally:
        os.environ.pop('MGPROTO_NO_FASTAUG')
        T._FASTAUG[:] = [None, False]
    assert float((a - b).abs().mean()) < 0.02
