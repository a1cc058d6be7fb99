"""Pure-PyTorch reference implementations of the prototype-math ops.

These are (a) the CPU execution path, (b) the fp32 oracle every HIP kernel is
parity-tested against, and (c) the mathematical spec for the kernels. Each op
documents the reference code it reproduces (file:line into /root/reference).

All ops are written vectorized (no per-class/per-sample Python loops — the
reference's loops at model.py:240-245, model.py:281 and memory.py:48 are
exactly the CPU serialization the rebuild removes).

Shapes (defaults): B=batch, H,W=latent grid, HW=H*W, N=B*HW, d=feature dim,
C=classes, K=prototypes per class, P=C*K, T=mining levels.

Layout convention: patch features are packed row-major as ``feat[N, d]`` with
row ``n = b*HW + hw``; per-patch scores as ``[N, P]`` with column
``p = c*K + k``. The reference's ``[B, C, K, H, W]`` tensors are views of
this layout.
"""

import math
from typing import Tuple

import torch

LOG_2PI = math.log(2.0 * math.pi)


# ---------------------------------------------------------------------------
# K1 — per-patch GMM log-likelihood (reference model.py:256-275)
# ---------------------------------------------------------------------------

def gmm_expand_params(means: torch.Tensor, covs: torch.Tensor, eps: float = 0.0
                      ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Fold (means, covs) into the GEMM form of the Gaussian log-density.

    With diagonal sigma, ``logN(x|mu,sigma)`` expands to
    ``bias[p] + x . A[p] + x^2 . B[p]`` where::

        A[p,j]   = mu[p,j] / sigma[p,j]^2
        B[p,j]   = -1 / (2 sigma[p,j]^2)
        bias[p]  = -d/2 log(2 pi) - sum_j log sigma[p,j] - sum_j mu[p,j]^2/(2 sigma[p,j]^2)

    Returns ``Wt [P, 2d]`` ([A | B] rows) and ``bias [P]``. This is the
    exact expression the HIP MFMA kernel evaluates (one [N,2d]x[2d,P] GEMM
    over [x, x^2]); the row-major [P, 2d] layout lets the kernel stage its
    B panel with coalesced reads and conflict-free LDS writes.
    """
    P = means.shape[0]
    d = means.shape[1]
    sig = covs + eps
    inv_var = 1.0 / (sig * sig)                      # [P, d]
    A = means * inv_var                              # [P, d]
    Bq = -0.5 * inv_var                              # [P, d]
    bias = (-0.5 * d * LOG_2PI
            - torch.log(sig).sum(dim=1)
            - 0.5 * (means * means * inv_var).sum(dim=1))   # [P]
    Wt = torch.cat([A, Bq], dim=1).contiguous()      # [P, 2d]
    return Wt, bias


def gmm_logprob(feat: torch.Tensor, means: torch.Tensor, covs: torch.Tensor,
                eps: float = 0.0) -> torch.Tensor:
    """log N(feat[n] | means[p], diag covs[p]) for all n, p  ->  [N, P].

    Reference: model.py:256-275 (`compute_log_prob`), which evaluates the
    direct (x-mu)^2/sigma^2 broadcast in n_block x c_block groups. Here the
    quadratic is expanded into a single GEMM (no [N,P,d] temporary), which is
    what makes the op MFMA-shaped. means/covs are treated as constants
    (the reference detaches them at model.py:264-265); gradients flow to
    ``feat`` only.
    """
    means = means.reshape(-1, means.shape[-1]).detach()
    covs = covs.reshape(-1, covs.shape[-1]).detach()
    Wt, bias = gmm_expand_params(means, covs, eps)
    x2 = torch.cat([feat, feat * feat], dim=1)       # [N, 2d]
    return x2 @ Wt.t() + bias


def gmm_logprob_direct(feat: torch.Tensor, means: torch.Tensor,
                       covs: torch.Tensor, eps: float = 0.0) -> torch.Tensor:
    """Direct broadcast evaluation (the reference's exact expression).

    Used only as a numerics oracle in tests; O(N*P*d) memory per block.
    """
    means = means.reshape(-1, means.shape[-1]).detach()
    covs = covs.reshape(-1, covs.shape[-1]).detach()
    d = feat.shape[1]
    out = torch.empty(feat.shape[0], means.shape[0],
                      device=feat.device, dtype=feat.dtype)
    blk = max(1, (1 << 22) // max(1, means.shape[0] * d))
    const = -0.5 * d * LOG_2PI - torch.log(covs + eps).sum(-1)   # [P]
    for n0 in range(0, feat.shape[0], blk):
        diff = feat[n0:n0 + blk, None, :] - means                 # [blk, P, d]
        out[n0:n0 + blk] = const - 0.5 * (diff / (covs + eps)).pow(2).sum(-1)
    return out


def gmm_probs(feat: torch.Tensor, means: torch.Tensor, covs: torch.Tensor,
              eps: float = 0.0) -> torch.Tensor:
    """exp(gmm_logprob) — the per-patch mixture-component likelihoods [N, P].

    Reference: model.py:213-215 (`compute_log_prob(...).exp()`). The HIP
    kernel fuses the exp into the GEMM epilogue.
    """
    return torch.exp(gmm_logprob(feat, means, covs, eps))


# ---------------------------------------------------------------------------
# K2 — top-T spatial mining (reference model.py:188-206)
# ---------------------------------------------------------------------------

def topk_hw(probs: torch.Tensor, T: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per (image, prototype) top-T over the spatial axis.

    ``probs``: [B, HW, P]  ->  (values [B, P, T], indices [B, P, T]), values
    sorted descending. Reference: model.py:190 (`torch.topk` over HW of the
    [B, P, HW] view). Ties: torch.topk order (first occurrence wins) — the
    HIP kernel matches lowest-index-wins tie-breaking.
    """
    vals, idx = torch.topk(probs, T, dim=1)          # [B, T, P] over HW axis
    return vals.permute(0, 2, 1), idx.permute(0, 2, 1)


def gather_patch_features(feat: torch.Tensor, idx: torch.Tensor, B: int,
                          HW: int) -> torch.Tensor:
    """Gather packed patch features at per-(b,p) spatial indices.

    ``feat``: [B*HW, d] packed; ``idx``: [B, P] (or [B, P, T]) of hw indices.
    Returns [B, P, d] (or [B, P, T, d]). Reference: model.py:197-204's
    T-times-replicated torch.gather loop, done as one flat index here.
    """
    d = feat.shape[1]
    base = (torch.arange(B, device=feat.device) * HW).view(B, *([1] * (idx.dim() - 1)))
    flat = (idx + base).reshape(-1)                  # [B*P(*T)]
    out = feat.index_select(0, flat)
    return out.reshape(*idx.shape, d)


def mask_wrong_class(vals: torch.Tensor, gt: torch.Tensor,
                     class_identity: torch.Tensor) -> torch.Tensor:
    """For mining levels k>=1, wrong-class prototypes keep their level-0 value.

    ``vals``: [B, P, T]; ``gt``: [B]; ``class_identity``: [P, C] one-hot.
    Reference: model.py:218-221 (in-place masked copy). Implemented
    out-of-place with torch.where — the autograd semantics (grad of a
    wrong-class level-k slot flows to its level-0 slot) are identical.
    """
    wrong = (1.0 - class_identity[:, gt].t()).bool()          # [B, P]
    lvl0 = vals[:, :, :1]
    keep0 = wrong.unsqueeze(-1)                                # [B, P, 1]
    out = torch.where(keep0, lvl0.expand_as(vals), vals)
    # level 0 itself is never masked
    return torch.cat([vals[:, :, :1], out[:, :, 1:]], dim=2)


# ---------------------------------------------------------------------------
# K4 — mixture head (reference model.py:54-74, :222)
# ---------------------------------------------------------------------------

def mixture_head(vals: torch.Tensor, weight: torch.Tensor, C: int, K: int
                 ) -> torch.Tensor:
    """Per-level class mixture likelihood: [B, P, T] x pi -> [B, C, T].

    ``weight`` is the NonNegLinear weight [C, P] whose class-masked entries
    are exactly zero (asserted by NonNegLinear). Because W[c, p] = 0 unless
    p // K == c, F.linear reduces to a per-class weighted sum over its own K
    prototypes — computed here as an einsum over the [C, K] diagonal block
    (10x less work than the dense [B,P]x[P,C] GEMM; exact, not approximate).
    Reference: model.py:222 (a Python loop of F.linear over T levels).
    """
    B, P, T = vals.shape
    diag = torch.arange(C, device=weight.device)  # device-side: capture-safe
    pi = weight.view(C, C, K)[diag, diag]                          # [C, K]
    v = vals.view(B, C, K, T)
    return torch.einsum('bckt,ck->bct', v, pi)


# ---------------------------------------------------------------------------
# K5 — deduplicated memory-bank candidate extraction (reference model.py:228-252)
# ---------------------------------------------------------------------------

def enqueue_candidates(feat: torch.Tensor, top1_idx: torch.Tensor,
                       gt: torch.Tensor, C: int, K: int, HW: int
                       ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per-sample unique top-1 patch features of the GT class's prototypes.

    ``feat``: [B*HW, d] packed patch features; ``top1_idx``: [B, P] top-1 hw
    index per prototype; ``gt``: [B]. Returns (features [M, d], labels [M])
    ordered (class asc, batch index asc, patch index asc) — the same
    multiset and the same per-(class, sample) ascending-patch-index order the
    reference produces (model.py:230-249: classes via gt.unique() [sorted],
    samples in batch order, patch indices via torch.unique [sorted]).

    The reference's per-sample double Python loop (model.py:240-245) is the
    K5 hot spot; this is one sort + mask with STATIC [B*K] output shape and
    no host sync (hipGraph-capturable): duplicate rows carry the sentinel
    label C, which MemoryBank.push discards device-side.
    """
    B = gt.shape[0]
    device = feat.device
    ar = torch.arange(B, device=device)
    # per sample: the K top-1 indices of its GT class's prototypes
    own = top1_idx.view(B, C, K)[ar, gt]                     # [B, K]
    # dedup within each row, keeping ascending patch-index order
    s_idx, _ = torch.sort(own, dim=1)                         # [B, K]
    first = torch.ones_like(s_idx, dtype=torch.bool)
    first[:, 1:] = s_idx[:, 1:] != s_idx[:, :-1]
    # order rows by (class, batch idx): stable sort of gt
    order = torch.argsort(gt, stable=True)                    # [B]
    s_idx = s_idx[order]
    first = first[order]
    labels = gt[order].unsqueeze(1).expand(B, K)
    lab = torch.where(first, labels,
                      torch.full_like(labels, C)).reshape(-1)  # [B*K]
    b_of = order.unsqueeze(1).expand(B, K).reshape(-1)
    rows = b_of * HW + s_idx.reshape(-1)
    return feat.index_select(0, rows), lab


# ---------------------------------------------------------------------------
# K6/K7 — batched EM over dirty classes (reference model.py:277-401)
# ---------------------------------------------------------------------------

def em_e_step(x: torch.Tensor, means: torch.Tensor, covs: torch.Tensor,
              pi: torch.Tensor, eps: float = 1e-10
              ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Batched E-step over G classes.

    ``x``: [G, N, d] memory features; ``means``/``covs``: [G, K, d];
    ``pi``: [G, K]. Returns (weighted_log_prob [G, N, K], log_resp [G, N, K]).
    Reference: model.py:303-336 (`_e_step` + `_estimate_log_prob`, run per
    class in a Python loop there; batched over classes here).

    Uses the same quadratic-expansion GEMM form as K1 (gmm_expand_params):
    a [G,N,2d]x[G,2d,K] bmm instead of the [G,N,K,d] broadcast temporary
    (which is ~4 GB at the default shapes and was the EM hot spot).
    """
    d = x.shape[-1]
    sig = covs + eps                                          # [G, K, d]
    inv_var = 1.0 / (sig * sig)
    A = (means * inv_var).transpose(1, 2)                     # [G, d, K]
    Bq = (-0.5 * inv_var).transpose(1, 2)                     # [G, d, K]
    bias = (-0.5 * d * LOG_2PI
            - torch.log(sig).sum(-1)
            - 0.5 * (means * means * inv_var).sum(-1))        # [G, K]
    x2 = torch.cat([x, x * x], dim=2)                         # [G, N, 2d]
    W = torch.cat([A, Bq], dim=1)                             # [G, 2d, K]
    log_prob = torch.baddbmm(bias.unsqueeze(1), x2, W)        # [G, N, K]
    wlp = log_prob + torch.log(pi + eps).unsqueeze(1)         # [G, N, K]
    log_norm = torch.logsumexp(wlp, dim=2, keepdim=True)
    return wlp, wlp - log_norm


def em_m_step_grads(x: torch.Tensor, log_resp: torch.Tensor, wlp: torch.Tensor,
                    means: torch.Tensor, covs: torch.Tensor,
                    alpha: float = 0.1, lamda: float = 1.0,
                    eps: float = 1e-10
                    ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Closed-form M-step: (grad wrt means [G, K, d], unnormalized pi [G, K]).

    Reproduces the gradient the reference obtains via autograd at
    model.py:367-397 (`_m_step_diversified`): the responsibility-weighted
    NLL term plus the lamda-weighted pairwise diversity repulsion
    ``sum_{i != j} exp(-||mu_i - mu_j||^2) / (K^2 - K)``; sigma and pi are
    frozen inside. Derivation::

        L_nll  = -(1/N) sum_n sum_k resp[n,k] * ll[n,k]
        dL/dmu[k,j] = -(1/N) sum_n resp[n,k] * (x[n,j]-mu[k,j]) / sigma[k,j]^2
        dDiv/dmu[i] = -(4/S) sum_{b != i} exp(-d_ib) (mu[i]-mu[b]),  S = K^2-K
    """
    G, N, K = log_resp.shape
    resp = torch.exp(log_resp)
    resp = (resp + alpha) / (resp + alpha).sum(dim=2, keepdim=True)   # smoothing
    pi_unnorm = resp.sum(dim=1) + eps                                  # [G, K]

    sig = covs + eps
    inv_var = 1.0 / (sig * sig)                                        # [G, K, d]
    # -(1/N) * sum_n resp[n,k] (x[n,j]-mu[k,j]) / sigma^2
    rx = torch.einsum('gnk,gnd->gkd', resp, x)                         # [G, K, d]
    rsum = resp.sum(dim=1)                                             # [G, K]
    grad_nll = -(rx - rsum.unsqueeze(-1) * means) * inv_var / N        # [G, K, d]

    # diversity repulsion
    diff = means.unsqueeze(2) - means.unsqueeze(1)                     # [G, K, K, d]
    dist = diff.pow(2).sum(-1)                                         # [G, K, K]
    w = torch.exp(-dist)
    eye = torch.eye(K, device=x.device, dtype=torch.bool)
    w = w.masked_fill(eye, 0.0)
    # K=1: no pairs to repel — the reference's 0/0 here would NaN the means
    S = max(K * K - K, 1)
    grad_div = -(4.0 / S) * torch.einsum('gik,gikd->gid', w, diff)     # [G, K, d]

    return grad_nll + lamda * grad_div, pi_unnorm


# ---------------------------------------------------------------------------
# K8 — push distance argmin (reference push.py:125-158)
# ---------------------------------------------------------------------------

def argmin_hw(probs: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per (image, prototype) nearest patch: distances are -probs, so the
    argmin of distance is the argmax of probs over HW.

    ``probs``: [B, HW, P] -> (min_dist [B, P], hw_index [B, P]).
    Reference: push.py:134-135 (np.argmin on the CPU copy); here it stays on
    device. np.argmin returns the first minimum; torch.max's index choice on
    ties matches lowest-index on contiguous CPU input, and the HIP kernel
    enforces lowest-index-wins.
    """
    vals, idx = probs.max(dim=1)                              # over HW
    return -vals, idx


# ---------------------------------------------------------------------------
# Aux — Proxy-Anchor loss pieces (reference utils/losses.py:41-61) live in
# mgproto_amd.losses; they are small GEMMs and stay on the PyTorch path.
# ---------------------------------------------------------------------------
