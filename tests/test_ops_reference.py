"""Parity tests of the vectorized ops against the reference math.

The oracles here are literal evaluations of the formulas the reference code
computes (cited per test); the framework implementations are the vectorized /
GEMM-shaped rewrites in mgproto_amd.ops.reference. On GPU, a separate suite
(test_ops_gpu.py) compares the HIP kernels against these same oracles.
"""

import math

import pytest
import torch
import torch.nn.functional as F

from mgproto_amd.ops import reference as R


def make_gmm(N=64, P=30, d=16, seed=0):
    g = torch.Generator().manual_seed(seed)
    feat = F.normalize(torch.randn(N, d, generator=g), dim=1)
    means = F.normalize(torch.rand(P, d, generator=g), dim=1)
    covs = torch.full((P, d), 1 / math.sqrt(2 * math.pi))
    return feat, means, covs


class TestGMMLogprob:
    def test_gemm_matches_direct(self):
        feat, means, covs = make_gmm()
        lp_gemm = R.gmm_logprob(feat, means, covs)
        lp_direct = R.gmm_logprob_direct(feat, means, covs)
        assert torch.allclose(lp_gemm, lp_direct, atol=1e-4, rtol=1e-5)

    def test_direct_matches_formula(self):
        # reference model.py:272: -d/2 log2pi - sum log sigma - 0.5 sum((x-mu)/sigma)^2
        feat, means, covs = make_gmm(N=8, P=5, d=4)
        lp = R.gmm_logprob_direct(feat, means, covs)
        for n in range(8):
            for p in range(5):
                diff = (feat[n] - means[p]) / covs[p]
                want = (-0.5 * 4 * math.log(2 * math.pi)
                        - torch.log(covs[p]).sum()
                        - 0.5 * diff.pow(2).sum())
                assert torch.allclose(lp[n, p], want, atol=1e-5)

    def test_nonuniform_sigma(self):
        feat, means, covs = make_gmm()
        covs = covs * (0.5 + torch.rand_like(covs))
        lp_gemm = R.gmm_logprob(feat, means, covs)
        lp_direct = R.gmm_logprob_direct(feat, means, covs)
        assert torch.allclose(lp_gemm, lp_direct, atol=1e-4, rtol=1e-5)

    def test_grad_flows_to_feat_only(self):
        feat, means, covs = make_gmm()
        feat.requires_grad_(True)
        means = means.clone().requires_grad_(True)
        out = R.gmm_probs(feat, means, covs)
        out.sum().backward()
        assert feat.grad is not None and feat.grad.abs().sum() > 0
        assert means.grad is None  # detached inside (reference model.py:264)

    def test_grad_matches_direct_path(self):
        feat, means, covs = make_gmm()
        f1 = feat.clone().requires_grad_(True)
        f2 = feat.clone().requires_grad_(True)
        R.gmm_probs(f1, means, covs).sum().backward()
        torch.exp(R.gmm_logprob_direct(f2, means, covs)).sum().backward()
        assert torch.allclose(f1.grad, f2.grad, atol=1e-4, rtol=1e-4)


class TestTopkMining:
    def test_topk_matches_reference_layout(self):
        # reference model.py:189-190: topk over HW of [B, P, HW]
        B, HW, P, T = 3, 49, 20, 5
        probs = torch.rand(B, HW, P)
        vals, idx = R.topk_hw(probs, T)
        ref_vals, ref_idx = torch.topk(probs.permute(0, 2, 1), T, dim=2)
        assert torch.equal(vals, ref_vals)
        assert torch.equal(idx, ref_idx)

    def test_gather_patch_features(self):
        B, HW, P, d = 2, 9, 6, 4
        feat = torch.randn(B * HW, d)
        idx = torch.randint(0, HW, (B, P))
        out = R.gather_patch_features(feat, idx, B, HW)
        for b in range(B):
            for p in range(P):
                assert torch.equal(out[b, p], feat[b * HW + idx[b, p]])

    def test_mask_wrong_class(self):
        # reference model.py:218-221
        B, C, K, T = 4, 5, 2, 3
        P = C * K
        identity = torch.zeros(P, C)
        for j in range(P):
            identity[j, j // K] = 1
        vals = torch.rand(B, P, T)
        gt = torch.tensor([0, 2, 2, 4])
        out = R.mask_wrong_class(vals, gt, identity)
        wrong = (1 - identity[:, gt].t()).bool()
        for b in range(B):
            for p in range(P):
                for k in range(T):
                    if k >= 1 and wrong[b, p]:
                        assert out[b, p, k] == vals[b, p, 0]
                    else:
                        assert out[b, p, k] == vals[b, p, k]

    def test_mixture_head_matches_linear(self):
        # reference model.py:222: F.linear with class-masked weight, per level
        B, C, K, T = 4, 5, 3, 2
        P = C * K
        identity = torch.zeros(P, C)
        for j in range(P):
            identity[j, j // K] = 1
        w = torch.rand(C, P) * identity.t()  # class-masked weights
        vals = torch.rand(B, P, T)
        out = R.mixture_head(vals, w, C, K)
        ref = torch.stack([F.linear(vals[:, :, k], w) for k in range(T)], dim=2)
        assert torch.allclose(out, ref, atol=1e-6)


class TestEnqueue:
    def test_matches_python_loop_oracle(self):
        # oracle = reference model.py:228-249 semantics: classes ascending,
        # samples in batch order, unique patch indices ascending
        B, C, K, HW, d = 6, 4, 3, 25, 8
        P = C * K
        feat = torch.randn(B * HW, d)
        top1 = torch.randint(0, HW, (B, P))
        gt = torch.tensor([2, 0, 2, 1, 0, 2])

        feats, labs = R.enqueue_candidates(feat, top1, gt, C, K, HW)

        want_feats, want_labs = [], []
        for c in sorted(gt.unique().tolist()):
            for b in range(B):
                if gt[b] != c:
                    continue
                own = top1[b, c * K:(c + 1) * K]
                for v in torch.unique(own):  # sorted ascending
                    want_feats.append(feat[b * HW + v])
                    want_labs.append(c)
        want_feats = torch.stack(want_feats)
        want_labs = torch.tensor(want_labs)
        assert torch.equal(labs, want_labs)
        assert torch.allclose(feats, want_feats)


class TestEM:
    def _estimate_log_prob_oracle(self, x, mu, var, eps=1e-10):
        # reference model.py:323-336
        d = x.shape[-1]
        log_p = ((x.unsqueeze(1) - mu) / (var + eps)).pow(2).sum(-1)
        log_sigma = torch.log(var + eps).sum(-1)
        return -0.5 * d * math.log(2 * math.pi) - log_sigma - 0.5 * log_p

    def test_e_step_matches_oracle(self):
        # reference model.py:303-321 per class; ours batched over classes
        G, N, K, d = 3, 40, 5, 8
        x = F.normalize(torch.randn(G, N, d), dim=2)
        means = F.normalize(torch.rand(G, K, d), dim=2)
        covs = torch.full((G, K, d), 1 / math.sqrt(2 * math.pi))
        pi = torch.softmax(torch.rand(G, K), dim=1)
        wlp, log_resp = R.em_e_step(x, means, covs, pi)
        for g in range(G):
            lp = self._estimate_log_prob_oracle(x[g], means[g], covs[g])
            w = lp + torch.log(pi[g] + 1e-10)
            norm = torch.logsumexp(w, dim=1, keepdim=True)
            assert torch.allclose(wlp[g], w, atol=1e-5)
            assert torch.allclose(log_resp[g], w - norm, atol=1e-5)

    def test_m_step_grad_matches_autograd(self):
        # oracle: autograd through the reference's m-step loss
        # (model.py:387-393: resp-weighted NLL + diversity repulsion)
        G, N, K, d = 2, 30, 4, 6
        x = F.normalize(torch.randn(G, N, d), dim=2)
        means = F.normalize(torch.rand(G, K, d), dim=2)
        covs = torch.full((G, K, d), 1 / math.sqrt(2 * math.pi))
        pi = torch.softmax(torch.rand(G, K), dim=1)
        alpha, lamda = 0.1, 1.0

        wlp, log_resp = R.em_e_step(x, means, covs, pi)
        grad, pi_unnorm = R.em_m_step_grads(x, log_resp, wlp, means, covs,
                                            alpha=alpha, lamda=lamda)

        for g in range(G):
            mu = means[g].clone().requires_grad_(True)
            resp = torch.exp(log_resp[g])
            resp = (resp + alpha) / (resp + alpha).sum(1, keepdim=True)
            ll = self._estimate_log_prob_oracle(x[g], mu, covs[g]) \
                + torch.log(pi[g] + 1e-10)
            wll = -(resp.detach() * ll).sum(1).mean(0)
            pd = torch.sum((mu.unsqueeze(2) - mu.t().unsqueeze(0)) ** 2, dim=1)
            I = 1 - torch.eye(K)
            div = (torch.exp(-pd) * I).sum() / I.sum()
            (wll + lamda * div).backward()
            assert torch.allclose(grad[g], mu.grad, atol=1e-5, rtol=1e-4)
            # pi: sum of smoothed responsibilities + eps (model.py:385)
            assert torch.allclose(pi_unnorm[g], resp.sum(0) + 1e-10, atol=1e-6)


class TestArgmin:
    def test_argmin_matches_numpy_semantics(self):
        # reference push.py:134-135: np.argmin over the distance map
        B, HW, P = 3, 16, 7
        probs = torch.rand(B, HW, P)
        min_dist, idx = R.argmin_hw(probs)
        dist = -probs
        for b in range(B):
            for p in range(P):
                j = int(torch.argmin(dist[b, :, p]))
                assert idx[b, p] == j
                assert min_dist[b, p] == dist[b, j, p]


class TestDispatchWrappers:
    def test_cpu_dispatch_equals_reference(self):
        """ops.em_e_step / em_m_step_grads / enqueue_candidates wrappers
        route CPU tensors to the oracle regardless of the opt-in envs."""
        import os

        import mgproto_amd.ops as O
        os.environ['MGPROTO_HIP_EM'] = '1'
        os.environ['MGPROTO_HIP_ENQUEUE'] = '1'
        try:
            g = torch.Generator().manual_seed(0)
            x = F.normalize(torch.randn(2, 20, 8, generator=g), dim=2)
            means = F.normalize(torch.rand(2, 3, 8, generator=g), dim=2)
            covs = torch.full((2, 3, 8), 0.4)
            pi = torch.softmax(torch.rand(2, 3, generator=g), dim=1)
            for a, b in zip(O.em_e_step(x, means, covs, pi),
                            R.em_e_step(x, means, covs, pi)):
                assert torch.equal(a, b)
            wlp, lr = R.em_e_step(x, means, covs, pi)
            for a, b in zip(O.em_m_step_grads(x, lr, wlp, means, covs),
                            R.em_m_step_grads(x, lr, wlp, means, covs)):
                assert torch.equal(a, b)
            feat = torch.randn(4 * 6, 8, generator=g)
            top1 = torch.randint(0, 6, (4, 9), generator=g)
            gt = torch.randint(0, 3, (4,), generator=g)
            for a, b in zip(O.enqueue_candidates(feat, top1, gt, 3, 3, 6),
                            R.enqueue_candidates(feat, top1, gt, 3, 3, 6)):
                assert torch.equal(a, b)
        finally:
            os.environ.pop('MGPROTO_HIP_EM', None)
            os.environ.pop('MGPROTO_HIP_ENQUEUE', None)
