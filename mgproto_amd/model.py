"""MGProto model core — Mixture of Gaussian-distributed Prototypes.

Same public surface as the reference (``/root/reference/model.py``):
``construct_MGProto``, ``MGProto.forward(x, gt)``, ``compute_log_prob``,
``push_forward``, ``update_GMM``, ``prune_prototypes_topM``,
``set_last_layer_incorrect_connection``, ``NonNegLinear`` — with the compute
path re-designed for MI355X:

* the per-patch GMM log-likelihood (reference model.py:256-275) runs as one
  fused [N,2d]x[2d,P] MFMA GEMM + exp epilogue (``ops.gmm_scores``);
* top-T mining + feature gather (model.py:188-206) is a single device op,
  not a Python loop of T gathers;
* memory-bank enqueue (model.py:228-252) is a batched sort/scatter with no
  per-sample Python loops, routed through a pluggable ``_enqueue_fn`` so the
  distributed wrapper can make it DP-correct (the reference's in-forward
  replica writes are lost under DataParallel — SURVEY.md §2.2 C2);
* the EM update (model.py:277-401) is batched over all dirty classes with
  closed-form M-step gradients and an internal per-class Adam, instead of
  one autograd graph + optimizer step per class per loop.

Deliberate behavioural divergences (each rank-invariant by construction):
* EM/Adam: the reference drives a single Adam over the whole
  ``prototype_means`` with per-class zero-padded grads (model.py:395-397),
  so every step also decays *other* classes' Adam momentum. Here Adam state
  and step counts are per-class and only dirty classes move.
* ``NonNegLinear`` asserts (model.py:69) are debug-gated — they are
  host-device syncs in the hot loop.
"""

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import ops
from .models.conv1x1 import GemmConv2d
from .models import (resnet18_features, resnet34_features, resnet50_features,
                     resnet101_features, resnet152_features,
                     densenet121_features, densenet161_features,
                     densenet169_features, densenet201_features,
                     vgg11_features, vgg11_bn_features, vgg13_features,
                     vgg13_bn_features, vgg16_features, vgg16_bn_features,
                     vgg19_features, vgg19_bn_features)
from .utils.memory import MemoryBank
from .utils.receptive_field import compute_proto_layer_rf_info_v2

base_architecture_to_features = {
    'resnet18': resnet18_features,
    'resnet34': resnet34_features,
    'resnet50': resnet50_features,
    'resnet101': resnet101_features,
    'resnet152': resnet152_features,
    'densenet121': densenet121_features,
    'densenet161': densenet161_features,
    'densenet169': densenet169_features,
    'densenet201': densenet201_features,
    'vgg11': vgg11_features,
    'vgg11_bn': vgg11_bn_features,
    'vgg13': vgg13_features,
    'vgg13_bn': vgg13_bn_features,
    'vgg16': vgg16_features,
    'vgg16_bn': vgg16_bn_features,
    'vgg19': vgg19_features,
    'vgg19_bn': vgg19_bn_features,
}


def l2_normalize(x, dim):
    return F.normalize(x, p=2, dim=dim)


def momentum_update(old_value, new_value, momentum):
    return momentum * old_value + (1 - momentum) * new_value


class NonNegLinear(nn.Module):
    """Class-masked non-negative mixture weights (reference model.py:54-74).

    The weight [C, P] holds the mixture priors pi; entries outside a class's
    own K prototypes are exactly zero. The weight is *state* (EM-updated),
    not gradient-trained.
    """

    def __init__(self, in_features: int, out_features: int,
                 prototype_class_identity=None, device=None, dtype=None):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.prototype_class_identity = prototype_class_identity
        self.weight = nn.Parameter(
            torch.ones((out_features, in_features), device=device, dtype=dtype),
            requires_grad=False)
        self.debug_asserts = False

    def forward(self, input, prototypes_to_keep_with_negative=None):
        if self.debug_asserts:
            neg = 1 - torch.t(self.prototype_class_identity)
            assert torch.sum(self.weight.data[neg == 1]) == 0
            if prototypes_to_keep_with_negative is not None:
                assert torch.sum(
                    self.weight.data[prototypes_to_keep_with_negative == 0]) == 0
        return F.linear(input, self.weight, bias=None)


class MGProto(nn.Module):
    def __init__(self, features, img_size, prototype_shape,
                 proto_layer_rf_info, num_classes, init_weights=True,
                 prototype_activation_function='log',
                 add_on_layers_type='bottleneck',
                 sz_embedding=32,
                 mem_capacity=800,
                 mine_K=20):
        super().__init__()
        self.img_size = img_size
        self.prototype_shape = prototype_shape
        self.num_prototypes = prototype_shape[0]
        self.num_classes = num_classes
        self.epsilon = 1e-4
        self.prototype_activation_function = prototype_activation_function

        assert self.num_prototypes % self.num_classes == 0
        self.num_prototypes_per_class = self.num_prototypes // self.num_classes

        # one-hot [P, C] prototype -> class map (constant; not checkpointed,
        # matching the reference where it is a plain attribute)
        identity = torch.zeros(self.num_prototypes, self.num_classes)
        for j in range(self.num_prototypes):
            identity[j, j // self.num_prototypes_per_class] = 1
        self.register_buffer('prototype_class_identity', identity,
                             persistent=False)

        self.proto_layer_rf_info = proto_layer_rf_info
        self.features = features

        features_name = str(self.features).upper()
        if features_name.startswith('VGG') or features_name.startswith('RES'):
            first_add_on_layer_in_channels = \
                [m for m in features.modules() if isinstance(m, nn.Conv2d)][-1].out_channels
        elif features_name.startswith('DENSE'):
            first_add_on_layer_in_channels = \
                [m for m in features.modules() if isinstance(m, nn.BatchNorm2d)][-1].num_features
        else:
            raise Exception('other base architecture NOT implemented')

        d = self.prototype_shape[1]
        if add_on_layers_type == 'bottleneck':
            add_on_layers = []
            current_in = first_add_on_layer_in_channels
            while (current_in > d) or (len(add_on_layers) == 0):
                current_out = max(d, (current_in // 2))
                add_on_layers.append(GemmConv2d(current_in, current_out, kernel_size=1))
                add_on_layers.append(nn.ReLU())
                add_on_layers.append(GemmConv2d(current_out, current_out, kernel_size=1))
                if current_out > d:
                    add_on_layers.append(nn.ReLU())
                else:
                    assert current_out == d
                    add_on_layers.append(nn.Sigmoid())
                current_in = current_in // 2
            self.add_on_layers = nn.Sequential(*add_on_layers)
        elif add_on_layers_type == 'regular_upsample':
            # the reference's commented-out R50/iNat variant (model.py:138):
            # 2x bilinear upsample -> 28x28 latent grid at 224 input.
            # The upsample COMMUTES exactly with the (linear, no-activation)
            # 1x1 convs, so it runs LAST: the convs see the 14x14 grid (4x
            # less work) and the upsample moves [B,d,..] instead of
            # [B,2048,..] tensors (32x less traffic; its backward alone was
            # 1.7 ms/step the other way — profiles/).
            from .models.upsample import Upsample2x
            self.add_on_layers = nn.Sequential(
                GemmConv2d(first_add_on_layer_in_channels, d, kernel_size=1),
                GemmConv2d(d, d, kernel_size=1),
                # exact nn.Upsample(2, bilinear) semantics; HIP gather
                # backward replaces torch's 1.69 ms/step atomic scatter
                Upsample2x(),
            )
        else:  # 'regular'
            self.add_on_layers = nn.Sequential(
                GemmConv2d(first_add_on_layer_in_channels, d, kernel_size=1),
                GemmConv2d(d, d, kernel_size=1),
            )

        self.gap = nn.AdaptiveAvgPool2d(1)
        self.embedding = nn.Linear(first_add_on_layer_in_channels, sz_embedding)

        self.prototype_means = nn.Parameter(
            torch.rand(self.num_classes, self.num_prototypes_per_class, d),
            requires_grad=True)
        self.prototype_means.data.copy_(l2_normalize(self.prototype_means.data, dim=2))

        self.init_sigma = 1 / math.sqrt(2 * math.pi)
        self.prototype_covs = nn.Parameter(
            torch.ones(self.num_classes, self.num_prototypes_per_class, d)
            * self.init_sigma, requires_grad=False)

        self.last_layer = NonNegLinear(self.num_prototypes, self.num_classes,
                                       prototype_class_identity=identity)

        if init_weights:
            self.initialize_weights()

        self.mine_T = mine_K
        self.capacity_pc = mem_capacity
        self.queue = MemoryBank(self.num_classes, d,
                                self.capacity_pc * self.num_classes, mode='all')
        # dirty-class flags with one extra sentinel slot so fixed-size
        # enqueue batches (labels == C for padding) index it without any
        # masked compaction (hipGraph-capturable)
        self.register_buffer('_memory_updated',
                             torch.zeros(self.num_classes + 1, dtype=torch.bool),
                             persistent=False)
        self.iteration_counter = nn.Parameter(torch.zeros(1), requires_grad=False)

        # EM hyper-parameters (reference model.py:171-174)
        self.update_interval = 1
        self.num_em_loop = 3
        self.alpha = 0.1
        self.tau = 0.990
        self.lamda = 1.0

        # internal per-class Adam over prototype_means (replaces the external
        # optimizer the reference hands into the model, model.py:395-397)
        self.prototype_lr = 3e-3
        self.adam_betas = (0.9, 0.999)
        self.adam_eps = 1e-8
        self.register_buffer('_em_exp_avg', torch.zeros_like(self.prototype_means),
                             persistent=False)
        self.register_buffer('_em_exp_avg_sq', torch.zeros_like(self.prototype_means),
                             persistent=False)
        self.register_buffer('_em_step', torch.zeros(self.num_classes,
                                                     dtype=torch.int64),
                             persistent=False)
        self.prototype_optimizer = None  # API-compat slot; unused internally
        # NOTE: the _em_* Adam buffers are non-persistent so the exported
        # state_dict stays exactly the reference checkpoint layout
        # (tests/test_checkpoint_interop.py); the training resume path
        # carries them separately via em_state_dict()/load_em_state().

        # pluggable enqueue: the distributed wrapper replaces this with an
        # all-gather + replicated push (parallel/state_sync.py)
        self._enqueue_fn = self._local_enqueue

        self.prototypes_to_keep = None
        self.prototypes_to_keep_with_negative = None

    # ------------------------------------------------------------- features
    def conv_features(self, x):
        x = self.features(x)
        x_add = self.add_on_layers(x)
        x_avg = self.gap(x).flatten(1)
        x_embed = l2_normalize(self.embedding(x_avg), dim=1)
        return x_add, x_embed

    # -------------------------------------------------------------- forward
    def forward(self, x, gt):
        base_feature, x_auxiliary = self.conv_features(x)
        # prototype math runs fp32 even when the backbone is under bf16
        # autocast (the engine autocasts only the conv stack)
        with torch.autocast(device_type=base_feature.device.type, enabled=False):
            return self._prototype_forward(base_feature, x_auxiliary, gt)

    def _prototype_forward(self, base_feature, x_auxiliary, gt):
        base_feature = l2_normalize(base_feature.float(), dim=1)
        B, d, H, W = base_feature.shape
        HW = H * W
        P = self.num_prototypes
        C, K = self.num_classes, self.num_prototypes_per_class

        # packed [N, d] patch features, row n = b*HW + hw
        feat = base_feature.permute(0, 2, 3, 1).reshape(B * HW, d)

        probs = ops.gmm_scores(feat, self.prototype_means, self.prototype_covs,
                               apply_exp=True)                     # [N, P]
        # mining pool cannot exceed the latent grid (e.g. mine_T=20 on a
        # small-image 4x4 grid — the reference's topk would error out)
        vals, idx = ops.topk_hw(probs.view(B, HW, P),
                                min(self.mine_T, HW))       # [B, P, T]

        if gt is not None:
            vals = ops.mask_wrong_class(vals, gt, self.prototype_class_identity)

        final_probs = ops.mixture_head(vals, self.last_layer.weight, C, K)  # [B, C, T]

        if gt is not None:
            with torch.no_grad():
                enq_feat, enq_lab = ops.enqueue_candidates(
                    feat.detach(), idx[:, :, 0], gt, C, K, HW)
                self._enqueue_fn(enq_feat, enq_lab)
            self.iteration_counter += 1

        return torch.log(final_probs), x_auxiliary

    @property
    def memory_updated_cls(self):
        """[C] dirty flags view (reference model.py:167 surface)."""
        return self._memory_updated[:self.num_classes]

    def _local_enqueue(self, feats, labels):
        self.queue.push(feats, labels)
        # index_fill_ keeps the scalar on device (index_put_ with a Python
        # bool uploads a host tensor and breaks graph capture)
        self._memory_updated.index_fill_(0, labels, True)

    # --------------------------------------------------- log-prob API parity
    def compute_log_prob(self, _fea, n_block: int = 4, c_block: int = 1,
                         eps: float = 0.0):
        """[N, d] -> [N, C, K] patch x prototype log-likelihoods.

        Reference model.py:256-275. n_block/c_block are accepted for API
        compatibility; the fused GEMM needs no blocking.
        """
        lp = ops.gmm_scores(_fea, self.prototype_means, self.prototype_covs,
                            apply_exp=False, eps=eps)
        return lp.view(_fea.shape[0], self.num_classes,
                       self.num_prototypes_per_class)

    # ------------------------------------------------------------------ push
    def push_forward(self, x):
        """Returns (l2-normalized conv features [B,d,H,W], distances [B,P,H,W]).

        Reference model.py:429-438: distances = -exp(log_prob).
        """
        base_feature, _ = self.conv_features(x)
        with torch.autocast(device_type=base_feature.device.type, enabled=False):
            base_feature = l2_normalize(base_feature.float(), dim=1)
            B, d, H, W = base_feature.shape
            feat = base_feature.permute(0, 2, 3, 1).reshape(B * H * W, d)
            probs = ops.gmm_scores(feat, self.prototype_means,
                                   self.prototype_covs, apply_exp=True)
        distances = -probs.view(B, H * W, self.num_prototypes) \
                          .permute(0, 2, 1).reshape(B, self.num_prototypes, H, W)
        return base_feature, distances

    def em_state_dict(self) -> dict:
        """Per-class EM Adam state for full resume. Kept OUT of the module
        state_dict so checkpoints stay byte-compatible with the reference
        layout; a resumed run restores it via load_em_state so the
        prototype trajectory continues instead of restarting from zeroed
        moments (the reference's external prototype optimizer state was
        never saved at all, reference utils/save.py:5-12)."""
        return {'exp_avg': self._em_exp_avg.detach().cpu().clone(),
                'exp_avg_sq': self._em_exp_avg_sq.detach().cpu().clone(),
                'step': self._em_step.detach().cpu().clone()}

    def load_em_state(self, state: dict):
        if not state:
            return
        with torch.no_grad():
            self._em_exp_avg.copy_(state['exp_avg'].to(self._em_exp_avg.device))
            self._em_exp_avg_sq.copy_(
                state['exp_avg_sq'].to(self._em_exp_avg_sq.device))
            self._em_step.copy_(state['step'].to(self._em_step.device))

    # -------------------------------------------------------------------- EM
    @torch.no_grad()
    def update_GMM(self):
        """Batched EM over all dirty+full classes (reference model.py:277-301).

        E-step + closed-form diversified M-step (ops.em_e_step /
        ops.em_m_step_grads), one fused launch set per EM loop for ALL
        classes at once instead of a per-class Python loop; per-class Adam
        on the means; pi momentum tau; priors written into the last layer.

        Shapes are STATIC ([C, cap, d] with an active-class mask) and there
        is no host synchronization, so the whole update runs asynchronously
        on a side HIP stream (EMRunner) and is hipGraph-capturable; inactive
        classes are computed and discarded (in steady state, with
        update_interval=1, every class in the batch history is dirty anyway).
        """
        dirty = self._memory_updated[:self.num_classes].clone()
        self._memory_updated.zero_()
        active = (dirty & self.queue.full_mask()).view(-1, 1)    # [C, 1]
        if not self.queue.mem.is_cuda and not bool(active.any()):
            return  # CPU path: skip the (cheap) masked compute entirely

        C, K = self.num_classes, self.num_prototypes_per_class
        x = self.queue.mem                                       # [C, cap, d]
        N = x.shape[1]
        w = self.last_layer.weight.data                          # [C, P]
        diag = torch.arange(C, device=w.device)
        pi_cur = w.view(C, C, K)[diag, diag]                     # [C, K]
        pi_old = pi_cur.clone()
        means = self.prototype_means.data.clone()                # [C, K, d]
        covs = self.prototype_covs.data

        for _ in range(self.num_em_loop):
            wlp, log_resp = ops.em_e_step(x, means, covs, pi_old)
            grad, pi_unnorm = ops.em_m_step_grads(
                x, log_resp, wlp, means, covs, alpha=self.alpha,
                lamda=self.lamda)
            means = self._em_adam_step(active, means, grad)
            pi = pi_unnorm / N
            pi_old = momentum_update(pi_old, pi, self.tau)

        act3 = active.unsqueeze(-1)                              # [C, 1, 1]
        self.prototype_means.data.copy_(
            torch.where(act3, means, self.prototype_means.data))
        w.view(C, C, K)[diag, diag] = torch.where(active, pi_old, pi_cur)

    def _em_adam_step(self, active, means, grad):
        """Masked per-class Adam: only active classes advance state/means."""
        b1, b2 = self.adam_betas
        act3 = active.unsqueeze(-1)                              # [C, 1, 1]
        self._em_step += active.view(-1).to(self._em_step.dtype)
        step = self._em_step.to(means.dtype).view(-1, 1, 1)
        m = b1 * self._em_exp_avg + (1 - b1) * grad
        v = b2 * self._em_exp_avg_sq + (1 - b2) * grad * grad
        self._em_exp_avg.copy_(torch.where(act3, m, self._em_exp_avg))
        self._em_exp_avg_sq.copy_(torch.where(act3, v, self._em_exp_avg_sq))
        # scalar**tensor stays on device (a torch.tensor(...) here would be a
        # pageable H2D copy and break hipGraph capture)
        bc1 = 1 - torch.pow(b1, step)
        bc2 = 1 - torch.pow(b2, step)
        denom = (v / bc2).sqrt() + self.adam_eps
        stepped = means - self.prototype_lr * (m / bc1) / denom
        return torch.where(act3, stepped, means)

    # ------------------------------------------- EM API parity (per class)
    # Single-class forms matching the reference's method surface
    # (model.py:303-427); update_GMM uses the batched ops versions.

    def _check_size(self, x):
        if len(x.size()) == 2:
            x = x.unsqueeze(1)
        return x

    def _estimate_log_prob(self, x, mu, var, eps=1e-10):
        """[n, 1, d] x [1, k, d] -> [n, k, 1] (reference model.py:323-336)."""
        x = self._check_size(x)
        d = x.shape[-1]
        log_p = ((x - mu) / (var + eps)).pow(2).sum(dim=2, keepdim=True)
        log_sigma = torch.log(var + eps).sum(dim=2, keepdim=True)
        return -0.5 * d * math.log(2 * math.pi) - log_sigma - 0.5 * log_p

    def _e_step(self, x, mu, var, pi, eps=1e-10):
        """(mean log-evidence, log-responsibilities) — reference :303-321."""
        x = self._check_size(x)
        wlp = self._estimate_log_prob(x, mu, var) + torch.log(pi + eps)
        log_norm = torch.logsumexp(wlp, dim=1, keepdim=True)
        return torch.mean(log_norm), wlp - log_norm

    def _m_step(self, x, log_resp, eps=1e-10):
        """Closed-form M-step (reference :338-365 — dead code there, kept
        for API parity; update_GMM uses _m_step-diversified semantics)."""
        x = self._check_size(x)
        resp = torch.exp(log_resp)
        resp = (resp + self.alpha) / (resp + self.alpha).sum(1, keepdim=True)
        pi = torch.sum(resp, dim=0, keepdim=True) + eps
        mu = torch.sum(resp * x, dim=0, keepdim=True) / pi
        x2 = (resp * x * x).sum(0, keepdim=True) / pi
        xmu = (resp * mu * x).sum(0, keepdim=True) / pi
        var = (x2 - 2 * xmu + mu * mu + eps).sqrt()
        return pi / x.shape[0], mu, var

    # --------------------------------------------------------------- scoring
    def _score(self, x, mu, var, pi, as_average=True, eps=1e-10):
        """Data log-likelihood under one class's mixture (reference :403-421)."""
        wlp, _ = ops.em_e_step(x.unsqueeze(0) if x.dim() == 2 else x,
                               mu if mu.dim() == 3 else mu.unsqueeze(0),
                               var if var.dim() == 3 else var.unsqueeze(0),
                               pi.reshape(1, -1), eps=eps)
        per_sample = torch.logsumexp(wlp, dim=2).squeeze(0)
        return per_sample.mean() if as_average else per_sample.squeeze()

    # ------------------------------------------------------------------ init
    def set_last_layer_incorrect_connection(self, incorrect_strength):
        pos = torch.t(self.prototype_class_identity)
        neg = 1 - pos
        correct = 1. / self.num_prototypes_per_class
        self.last_layer.weight.data.copy_(correct * pos + incorrect_strength * neg)

    def initialize_weights(self):
        for m in self.add_on_layers.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode='fan_out', nonlinearity='relu')
                if m.bias is not None:
                    nn.init.constant_(m.bias, 0)
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)
        nn.init.kaiming_normal_(self.embedding.weight, mode='fan_out')
        nn.init.constant_(self.embedding.bias, 0)
        self.set_last_layer_incorrect_connection(incorrect_strength=0.0)

    # --------------------------------------------------------------- pruning
    @torch.no_grad()
    def prune_prototypes_topM(self, top_M: int = 1):
        """Keep each class's top-M prototypes by prior (reference :467-482)."""
        w = self.last_layer.weight  # [C, P]
        pos_mask = torch.t(self.prototype_class_identity).to(w.device)
        proto_prior = w[pos_mask == 1].view(self.num_classes, -1)    # [C, K]
        threshold, _ = torch.topk(proto_prior, top_M, dim=1)
        thr = threshold[:, -1:]                                       # [C, 1]
        self.prototypes_to_keep = proto_prior >= thr                  # [C, K]
        self.prototypes_to_keep_with_negative = w >= thr              # [C, P]
        assert (self.prototypes_to_keep.sum(1) >= 1).all()
        w.data[self.prototypes_to_keep_with_negative == 0] = 0.0


def construct_MGProto(base_architecture, pretrained=True, img_size=224,
                      prototype_shape=(2000, 128, 1, 1), num_classes=200,
                      prototype_activation_function='log',
                      add_on_layers_type='bottleneck',
                      sz_embedding=32,
                      mem_capacity=1000,
                      mine_K=10):
    """Reference model.py:485-510."""
    features = base_architecture_to_features[base_architecture](pretrained=pretrained)
    layer_filter_sizes, layer_strides, layer_paddings = features.conv_info()
    proto_layer_rf_info = compute_proto_layer_rf_info_v2(
        img_size=img_size,
        layer_filter_sizes=layer_filter_sizes,
        layer_strides=layer_strides,
        layer_paddings=layer_paddings,
        prototype_kernel_size=prototype_shape[2])
    return MGProto(features=features,
                   img_size=img_size,
                   prototype_shape=prototype_shape,
                   proto_layer_rf_info=proto_layer_rf_info,
                   num_classes=num_classes,
                   init_weights=True,
                   prototype_activation_function=prototype_activation_function,
                   add_on_layers_type=add_on_layers_type,
                   sz_embedding=sz_embedding,
                   mem_capacity=mem_capacity,
                   mine_K=mine_K)
