"""Prototype-math ops with CPU (PyTorch) and MI355X (HIP/CDNA4) paths.

Dispatch policy:
* CPU tensors -> the pure-PyTorch reference implementations (the oracle).
* CUDA (ROCm) tensors -> the in-tree ``mgproto_hip`` extension (gfx950
  MFMA kernels). If the extension is missing on a GPU box this raises —
  a silent eager fallback would invalidate benchmarks. Set
  ``MGPROTO_EAGER_FALLBACK=1`` to explicitly allow the PyTorch path on GPU
  (used for A/B measurements only).
"""

import os
from typing import Optional, Tuple

import torch

from . import reference
from .reference import (gmm_expand_params, gmm_logprob_direct, mask_wrong_class,  # noqa: F401
                        mixture_head, gather_patch_features)

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from . import hip_loader
        _EXT = hip_loader.load()
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = f'{type(e).__name__}: {e}'
        _EXT = None
    return _EXT


def native_available() -> bool:
    return _load_extension() is not None


def _native_or_die():
    ext = _load_extension()
    if ext is None:
        if os.environ.get('MGPROTO_EAGER_FALLBACK') == '1':
            return None
        raise RuntimeError(
            'mgproto_hip extension is required for GPU execution but could not '
            f'be loaded ({_EXT_ERR}). Build it with '
            '`python -m mgproto_amd.ops.build` (or __graft_entry__.build()), '
            'or set MGPROTO_EAGER_FALLBACK=1 to explicitly allow the slow '
            'PyTorch path.')
    return ext


class _GMMScore(torch.autograd.Function):
    """Fused GMM score GEMM: out[n,p] = f(bias[p] + [x, x^2] @ W), f = exp or id.

    Forward/backward run on the gfx950 MFMA kernels for CUDA tensors; grads
    flow to ``feat`` only (means/covs are detached by the caller, matching
    reference model.py:264-265).
    """

    @staticmethod
    def forward(ctx, feat, Wt, bias, apply_exp):
        # Wt is [P, 2d] (the fwd kernel's B-panel layout); the bwd kernel
        # streams the [2d, P] transpose, materialized once per call (~1 MB)
        # B=1 push sweeps reach here with a VIEW (permute+reshape of
        # [1,d,H,W] is expressible without a copy) — kernels need packed rows
        feat = feat.contiguous()
        d = feat.shape[1]
        ext = _native_or_die() if feat.is_cuda else None
        if ext is not None and (d % 8 != 0 or d > 128):
            ext = None   # out of the kernel's envelope: hipBLASLt GEMM path
        ctx.used_ext = ext is not None
        if ext is not None:
            out = ext.gmm_fwd(feat, Wt, bias, apply_exp)
        else:
            out = torch.addmm(bias.unsqueeze(0),
                              torch.cat([feat, feat * feat], dim=1), Wt.t())
            if apply_exp:
                out = torch.exp(out)
        ctx.save_for_backward(feat, Wt, out if apply_exp else torch.empty(0))
        ctx.apply_exp = apply_exp
        return out

    @staticmethod
    def backward(ctx, grad_out):
        feat, Wt, probs = ctx.saved_tensors
        g = grad_out * probs if ctx.apply_exp else grad_out
        ext = _native_or_die() if (feat.is_cuda and ctx.used_ext) else None
        if ext is not None:
            P = Wt.shape[0]
            if P % 4:
                # the bwd kernel streams g rows with float4 loads; pad the
                # K dimension (zero columns contribute nothing). Reachable
                # with e.g. C=37 x K=10 (Pets config).
                pad = 4 - P % 4
                g = torch.nn.functional.pad(g, (0, pad))
                Wt = torch.nn.functional.pad(Wt, (0, 0, 0, pad))
            grad_feat = ext.gmm_bwd(g.contiguous(), feat,
                                    Wt.t().contiguous())
        else:
            gw = g @ Wt                          # [N, 2d]
            d = feat.shape[1]
            grad_feat = gw[:, :d] + 2.0 * feat * gw[:, d:]
        return grad_feat, None, None, None


class _GMMScoreUniform(torch.autograd.Function):
    """Reduced GMM score for FROZEN isotropic sigma (the default model:
    prototype_covs is initialized to 1/sqrt(2 pi) and never updated,
    reference model.py:151-152). Every x^2-column of W is then the constant
    cuni = -1/(2 sigma^2), so

        out[n,p] = f(bias[p] + x[n] . A[p] + cuni * ||x[n]||^2)

    — HALF the GEMM of the general form. Exact fp32; falls back to torch
    off-GPU."""

    @staticmethod
    def forward(ctx, feat, A, bias, cuni, apply_exp):
        feat = feat.contiguous()     # B=1 push sweeps pass strided views
        d = feat.shape[1]
        rn2 = (feat * feat).sum(dim=1).contiguous()
        ext = _native_or_die() if feat.is_cuda else None
        # d%16: the reduced kernel's MFMA macro-step is 16 wide over K=d
        # (the general kernel's K=2d is always a multiple of 16 for d%8)
        if ext is not None and (d % 16 != 0 or d > 128):
            ext = None
        ctx.used_ext = ext is not None
        if ext is not None:
            out = ext.gmm_fwd_uni(feat, A, bias, rn2, cuni, apply_exp)
        else:
            out = torch.addmm(bias.unsqueeze(0), feat, A.t()) \
                + cuni * rn2.unsqueeze(1)
            if apply_exp:
                out = torch.exp(out)
        ctx.save_for_backward(feat, A, out if apply_exp else torch.empty(0))
        ctx.apply_exp = apply_exp
        ctx.cuni = cuni
        return out

    @staticmethod
    def backward(ctx, grad_out):
        feat, A, probs = ctx.saved_tensors
        g = grad_out * probs if ctx.apply_exp else grad_out
        rs = g.sum(dim=1).contiguous()
        ext = _native_or_die() if (feat.is_cuda and ctx.used_ext) else None
        if ext is not None:
            P = A.shape[0]
            if P % 4:
                pad = 4 - P % 4
                g = torch.nn.functional.pad(g, (0, pad))
                A = torch.nn.functional.pad(A, (0, 0, 0, pad))
            grad_feat = ext.gmm_bwd_uni(g.contiguous(), feat,
                                        A.t().contiguous(), rs, ctx.cuni)
        else:
            grad_feat = g @ A + (2.0 * ctx.cuni) * feat * rs.unsqueeze(1)
        return grad_feat, None, None, None, None


# uniform-sigma detection cache, keyed on the MODEL's covs tensor
# (stable identity across calls, unlike the per-call reshape view): the
# one host sync happens on first sight of a covs tensor — during an eager
# warmup step — never inside a hipGraph capture/replay
_uniform_cache = {}


def _uniform_inv_var(covs: torch.Tensor):
    """inv_var scalar if ALL sigma entries are equal, else None."""
    key = (id(covs), covs._version, covs.data_ptr(), covs.shape)
    hit = _uniform_cache.get(key)
    if hit is not None:
        return hit[0]
    flat = covs.detach().reshape(-1).float()
    first = flat[0]
    uniform = bool((flat == first).all())
    val = float(1.0 / (first * first)) if uniform else None
    if len(_uniform_cache) > 64:     # bound: models come and go in tests
        _uniform_cache.clear()
    _uniform_cache[key] = (val,)
    return val


def gmm_scores(feat: torch.Tensor, means: torch.Tensor, covs: torch.Tensor,
               apply_exp: bool = True, eps: float = 0.0) -> torch.Tensor:
    """[N, d] patch features -> [N, P] Gaussian (log-)likelihoods.

    The hot K1 op (reference model.py:213-215). ``apply_exp=True`` returns
    probabilities (forward path); ``False`` returns log-probabilities
    (`compute_log_prob` API parity). Frozen-isotropic covariances (the
    default model) dispatch to the half-K reduced kernel; a general diag
    sigma keeps the full [x, x^2] GEMM (checkpoint parity)."""
    means2 = means.reshape(-1, means.shape[-1]).detach()
    covs2 = covs.reshape(-1, covs.shape[-1]).detach()
    if eps == 0.0 and os.environ.get('MGPROTO_NO_GMM_UNI') != '1':
        inv_var = _uniform_inv_var(covs)
        if inv_var is not None:
            import math
            m = means2.float()
            A = (m * inv_var).contiguous()                       # [P, d]
            d = m.shape[1]
            # -sum_j log sigma = +d/2 log(inv_var)  (sigma uniform)
            bias = (-0.5 * d * reference.LOG_2PI
                    + 0.5 * d * math.log(inv_var)
                    - 0.5 * inv_var * (m * m).sum(dim=1))        # [P]
            return _GMMScoreUniform.apply(feat, A, bias.contiguous(),
                                          -0.5 * inv_var, apply_exp)
    Wt, bias = gmm_expand_params(means2.float(), covs2.float(), eps)
    return _GMMScore.apply(feat, Wt, bias, apply_exp)


class _TopkHW(torch.autograd.Function):
    """Per-(image, prototype) top-T over the spatial axis, with indices."""

    @staticmethod
    def forward(ctx, probs, T):
        ext = _native_or_die() if probs.is_cuda else None
        if ext is not None and T > 32:
            ext = None   # kernel register-list limit; torch.topk on GPU
        if ext is not None:
            vals, idx = ext.topk_hw(probs, T)
        else:
            v, i = torch.topk(probs, T, dim=1)
            vals, idx = v.permute(0, 2, 1).contiguous(), i.permute(0, 2, 1).contiguous()
        ctx.save_for_backward(idx)
        ctx.hw = probs.shape[1]
        return vals, idx.to(torch.long)

    @staticmethod
    def backward(ctx, grad_vals, _grad_idx):
        (idx,) = ctx.saved_tensors
        B, P, T = idx.shape
        grad_probs = grad_vals.new_zeros(B, ctx.hw, P)
        # scatter-add grads back to their source patches
        grad_probs.scatter_add_(1, idx.to(torch.long).permute(0, 2, 1),
                                grad_vals.permute(0, 2, 1))
        return grad_probs, None


def topk_hw(probs: torch.Tensor, T: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """[B, HW, P] -> (values [B, P, T] desc-sorted, indices [B, P, T])."""
    return _TopkHW.apply(probs, T)


def enqueue_candidates(feat: torch.Tensor, top1_idx: torch.Tensor,
                       gt: torch.Tensor, C: int, K: int, HW: int):
    """Per-sample dedup of the GT class's top-1 patches.

    Opt-in HIP path (MGPROTO_HIP_ENQUEUE=1; round-2 validated: unit +
    in-situ parity green on MI355X, throughput within noise of the
    default — docs/ENVVARS.md);
    default = the batched torch sort/mask (reference.py), same output."""
    if (feat.is_cuda and os.environ.get('MGPROTO_HIP_ENQUEUE') == '1'
            and gt.shape[0] <= 1024 and K <= 32
            and _load_extension() is not None):
        ext = _load_extension()
        rows, lab = ext.enqueue_rows(top1_idx.contiguous(),
                                     gt.contiguous(), C, K, HW)
        return feat.index_select(0, rows), lab
    return reference.enqueue_candidates(feat, top1_idx, gt, C, K, HW)


def _em_hip_usable(x: torch.Tensor, K: int) -> bool:
    """K6/K7 HIP kernels: opt-in (MGPROTO_HIP_EM=1) until GPU-validated;
    the default EM path is the rocBLAS baddbmm form (reference.py)."""
    return (x.is_cuda and os.environ.get('MGPROTO_HIP_EM') == '1'
            and x.shape[-1] <= 128 and K <= 32
            and _load_extension() is not None)


def em_e_step(x: torch.Tensor, means: torch.Tensor, covs: torch.Tensor,
              pi: torch.Tensor, eps: float = 1e-10):
    """Batched EM E-step. Same semantics as the oracle
    (reference.py em_e_step == reference model.py:303-336)."""
    if not _em_hip_usable(x, means.shape[1]):
        return reference.em_e_step(x, means, covs, pi, eps=eps)
    ext = _load_extension()
    d = x.shape[-1]
    sig = (covs + eps).float()
    inv_var = 1.0 / (sig * sig)
    A = (means.float() * inv_var).contiguous()                   # [G, K, d]
    B = (-0.5 * inv_var).contiguous()
    bias = (-0.5 * d * reference.LOG_2PI
            - torch.log(sig).sum(-1)
            - 0.5 * (means.float() * means.float() * inv_var).sum(-1)
            + torch.log(pi.float() + eps)).contiguous()          # [G, K]
    wlp, logresp = ext.em_estep(x.float().contiguous(), A, B, bias)
    return wlp, logresp


def em_m_step_grads(x: torch.Tensor, log_resp: torch.Tensor,
                    wlp: torch.Tensor, means: torch.Tensor,
                    covs: torch.Tensor, alpha: float = 0.1,
                    lamda: float = 1.0, eps: float = 1e-10):
    """Batched EM M-step closed-form grads."""
    if not _em_hip_usable(x, means.shape[1]):
        return reference.em_m_step_grads(x, log_resp, wlp, means, covs,
                                         alpha=alpha, lamda=lamda, eps=eps)
    ext = _load_extension()
    grad, pi_unnorm = ext.em_mstep(x.float().contiguous(),
                                   log_resp.float().contiguous(),
                                   means.float().contiguous(),
                                   covs.float().contiguous(),
                                   alpha, lamda, eps)
    return grad, pi_unnorm


def argmax_hw(probs: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """[B, HW, P] -> (max values [B, P], hw indices [B, P]). No autograd
    (push path runs under no_grad). Lowest-index-wins on ties."""
    if probs.is_cuda:
        ext = _native_or_die()
        if ext is not None:
            vals, idx = ext.argmax_hw(probs)
            return vals, idx.to(torch.long)
    vals, idx = probs.max(dim=1)
    return vals, idx
