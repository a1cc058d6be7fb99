"""Auxiliary deep-metric-learning losses on the 32-d embedding head.

The reference (``/root/reference/utils/losses.py``) implements Proxy-Anchor
inline and wraps pytorch-metric-learning for the rest; that library is not
available here, so all six losses are self-contained implementations of the
published formulations. This also fixes the reference's driver bug where
selecting any aux loss other than Proxy_Anchor crashes (main.py:189 tests
``args.loss`` which does not exist).

These are tiny (B x C) GEMMs — they stay on the PyTorch-ROCm path
(measure first; not perf-critical).
"""

import torch
import torch.nn as nn
import torch.nn.functional as F


def l2_norm(x: torch.Tensor) -> torch.Tensor:
    return F.normalize(x, p=2, dim=1, eps=1e-6)


def binarize(T: torch.Tensor, nb_classes: int) -> torch.Tensor:
    return F.one_hot(T.long(), nb_classes).float()


class Proxy_Anchor(nn.Module):
    """Proxy-Anchor loss (Kim et al., CVPR 2020; reference losses.py:29-61)."""

    def __init__(self, nb_classes, sz_embed, mrg=0.1, beta=32):
        super().__init__()
        self.proxies = nn.Parameter(torch.randn(nb_classes, sz_embed))
        nn.init.kaiming_normal_(self.proxies, mode='fan_out')
        self.nb_classes = nb_classes
        self.sz_embed = sz_embed
        self.mrg = mrg
        self.beta = beta

    def forward(self, X, T):
        P = self.proxies
        cos = F.linear(l2_norm(X), l2_norm(P))               # [B, C]
        P_one_hot = binarize(T, self.nb_classes).to(cos.device)
        N_one_hot = 1 - P_one_hot

        pos_exp = torch.exp(-self.beta * (cos - self.mrg))
        neg_exp = torch.exp(self.beta * (cos + self.mrg))

        # device-side count of classes present in the batch (the reference's
        # torch.nonzero + len() is a host sync and blocks hipGraph capture)
        num_valid_proxies = (P_one_hot.sum(dim=0) != 0).sum().clamp(min=1)

        P_sim_sum = torch.where(P_one_hot == 1, pos_exp,
                                torch.zeros_like(pos_exp)).sum(dim=0)
        N_sim_sum = torch.where(N_one_hot == 1, neg_exp,
                                torch.zeros_like(neg_exp)).sum(dim=0)

        pos_term = torch.log(1 + P_sim_sum).sum() / num_valid_proxies
        neg_term = torch.log(1 + N_sim_sum).sum() / self.nb_classes
        return pos_term + neg_term


class Proxy_NCA(nn.Module):
    """ProxyNCA (Movshovitz-Attias et al., ICCV 2017)."""

    def __init__(self, nb_classes, sz_embed, scale=32):
        super().__init__()
        self.proxies = nn.Parameter(torch.randn(nb_classes, sz_embed) / 8)
        self.nb_classes = nb_classes
        self.scale = scale

    def forward(self, X, T):
        P = l2_norm(self.proxies) * self.scale
        X = l2_norm(X) * self.scale
        dist = torch.cdist(X, P) ** 2                        # [B, C]
        loss = F.cross_entropy(-dist, T.long())
        return loss


class MultiSimilarityLoss(nn.Module):
    """Multi-similarity loss (Wang et al., CVPR 2019) with in-batch pairs."""

    def __init__(self, thresh=0.5, epsilon=0.1, scale_pos=2.0, scale_neg=50.0):
        super().__init__()
        self.thresh = thresh
        self.epsilon = epsilon
        self.scale_pos = scale_pos
        self.scale_neg = scale_neg

    def forward(self, X, T):
        X = l2_norm(X)
        sim = X @ X.t()                                      # [B, B]
        B = X.shape[0]
        eye = torch.eye(B, dtype=torch.bool, device=X.device)
        pos_mask = (T.unsqueeze(0) == T.unsqueeze(1)) & ~eye
        neg_mask = (T.unsqueeze(0) != T.unsqueeze(1))

        loss = X.new_zeros(())
        n_valid = 0
        for i in range(B):
            pos_sim = sim[i][pos_mask[i]]
            neg_sim = sim[i][neg_mask[i]]
            if pos_sim.numel() == 0 or neg_sim.numel() == 0:
                continue
            # pair mining (the reference's MultiSimilarityMiner)
            neg_sel = neg_sim[neg_sim + self.epsilon > pos_sim.min()]
            pos_sel = pos_sim[pos_sim - self.epsilon < neg_sim.max()]
            if neg_sel.numel() == 0 or pos_sel.numel() == 0:
                continue
            pos_term = (1.0 / self.scale_pos) * torch.log(
                1 + torch.exp(-self.scale_pos * (pos_sel - self.thresh)).sum())
            neg_term = (1.0 / self.scale_neg) * torch.log(
                1 + torch.exp(self.scale_neg * (neg_sel - self.thresh)).sum())
            loss = loss + pos_term + neg_term
            n_valid += 1
        return loss / max(n_valid, 1)


class ContrastiveLoss(nn.Module):
    def __init__(self, margin=0.5, **kwargs):
        super().__init__()
        self.margin = margin

    def forward(self, X, T):
        X = l2_norm(X)
        dist = torch.cdist(X, X)
        B = X.shape[0]
        eye = torch.eye(B, dtype=torch.bool, device=X.device)
        pos_mask = (T.unsqueeze(0) == T.unsqueeze(1)) & ~eye
        neg_mask = (T.unsqueeze(0) != T.unsqueeze(1))
        pos_loss = dist[pos_mask].pow(2)
        neg_loss = F.relu(self.margin - dist[neg_mask]).pow(2)
        terms = []
        if pos_loss.numel():
            terms.append(pos_loss.mean())
        if neg_loss.numel():
            terms.append(neg_loss.mean())
        if not terms:
            return X.new_zeros(())
        return sum(terms)


class TripletLoss(nn.Module):
    """Batch-hard triplet loss with semihard-style margin filtering."""

    def __init__(self, margin=0.1, **kwargs):
        super().__init__()
        self.margin = margin

    def forward(self, X, T):
        X = l2_norm(X)
        dist = torch.cdist(X, X)
        B = X.shape[0]
        eye = torch.eye(B, dtype=torch.bool, device=X.device)
        pos_mask = (T.unsqueeze(0) == T.unsqueeze(1)) & ~eye
        neg_mask = (T.unsqueeze(0) != T.unsqueeze(1))
        INF = torch.finfo(dist.dtype).max
        hardest_pos = torch.where(pos_mask, dist, torch.zeros_like(dist)).max(1).values
        hardest_neg = torch.where(neg_mask, dist, torch.full_like(dist, INF)).min(1).values
        valid = pos_mask.any(1) & neg_mask.any(1)
        if not valid.any():
            return X.new_zeros(())
        losses = F.relu(hardest_pos - hardest_neg + self.margin)[valid]
        return losses.mean()


class NPairLoss(nn.Module):
    def __init__(self, l2_reg=0):
        super().__init__()
        self.l2_reg = l2_reg

    def forward(self, X, T):
        # one positive pair per class present at least twice
        classes, counts = torch.unique(T, return_counts=True)
        anchors, positives = [], []
        for c in classes[counts >= 2]:
            idx = torch.nonzero(T == c, as_tuple=False).flatten()
            anchors.append(idx[0])
            positives.append(idx[1])
        if not anchors:
            return X.new_zeros(())
        a = X[torch.stack(anchors)]
        p = X[torch.stack(positives)]
        logits = a @ p.t()                                    # [m, m]
        target = torch.arange(a.shape[0], device=X.device)
        loss = F.cross_entropy(logits, target)
        if self.l2_reg:
            loss = loss + self.l2_reg * (a.norm(dim=1).mean() + p.norm(dim=1).mean())
        return loss


AUX_LOSSES = {
    'Proxy_Anchor': Proxy_Anchor,
    'Proxy_NCA': Proxy_NCA,
    'MS': MultiSimilarityLoss,
    'Contrastive': ContrastiveLoss,
    'Triplet': TripletLoss,
    'NPair': NPairLoss,
}


def build_aux_loss(name: str, nb_classes: int, sz_embed: int, **kwargs) -> nn.Module:
    """Aux-loss factory (fixes reference main.py:187-198 arg handling)."""
    if name in ('Proxy_Anchor',):
        return Proxy_Anchor(nb_classes=nb_classes, sz_embed=sz_embed,
                            mrg=kwargs.get('mrg', 0.1), beta=kwargs.get('beta', 32))
    if name == 'Proxy_NCA':
        return Proxy_NCA(nb_classes=nb_classes, sz_embed=sz_embed)
    if name == 'MS':
        return MultiSimilarityLoss()
    if name == 'Contrastive':
        return ContrastiveLoss()
    if name == 'Triplet':
        return TripletLoss()
    if name == 'NPair':
        return NPairLoss()
    raise ValueError(f'unknown aux loss {name!r}; options: {sorted(AUX_LOSSES)}')
