"""Learning-dynamics sanity: on linearly separable synthetic images the
full MGProto step (CE + mining + aux + enqueue + EM) must actually learn.

This guards the end-to-end training semantics (loss wiring, EM updating
priors/means, masking) — a broken sign or a dead gradient path shows up
here even though all unit parities pass.
"""

import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader, Dataset

from mgproto_amd.model import construct_MGProto
from mgproto_amd.losses import build_aux_loss


class SeparableImages(Dataset):
    """Class c = base noise + strong class-specific spatial pattern."""

    def __init__(self, n, num_classes, img_size=32, seed=0):
        g = torch.Generator().manual_seed(seed)
        self.patterns = torch.randn(num_classes, 3, img_size, img_size,
                                    generator=g)
        self.n = n
        self.num_classes = num_classes
        self.img_size = img_size
        self.seed = seed

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        g = torch.Generator().manual_seed(self.seed * 77777 + i)
        label = i % self.num_classes
        img = 0.4 * torch.randn(3, self.img_size, self.img_size, generator=g) \
            + self.patterns[label]
        return img, label, i


def test_training_learns_separable_data():
    torch.manual_seed(0)
    C, K, d = 4, 2, 16
    model = construct_MGProto('resnet18', pretrained=False, img_size=32,
                              prototype_shape=(C * K, d, 1, 1), num_classes=C,
                              add_on_layers_type='regular', sz_embedding=8,
                              mem_capacity=8, mine_K=2)
    aux = build_aux_loss('Proxy_Anchor', nb_classes=C, sz_embed=8)
    opt = torch.optim.Adam([
        {'params': model.features.parameters(), 'lr': 3e-4},
        {'params': model.add_on_layers.parameters(), 'lr': 3e-3},
        {'params': aux.parameters(), 'lr': 1e-2},
    ])
    ds = SeparableImages(32, C)
    loader = DataLoader(ds, batch_size=16,
                        collate_fn=lambda b: (torch.stack([x[0] for x in b]),
                                              torch.tensor([x[1] for x in b])))
    model.train()
    losses = []
    for epoch in range(6):
        for img, gt in loader:
            out, emb = model(img, gt)
            loss = F.cross_entropy(out[:, :, 0], gt) \
                + 0.2 * sum(F.cross_entropy(out[:, :, k], gt)
                            for k in range(1, out.shape[2])) / (out.shape[2] - 1) \
                + 0.5 * aux(emb, gt)
            opt.zero_grad(set_to_none=True)
            loss.backward()
            opt.step()
            model.update_GMM()  # EM active every step (update_interval=1)
            losses.append(float(loss.detach()))

    # loss must drop substantially and accuracy must beat chance by a lot
    assert losses[-1] < 0.6 * losses[0], (losses[0], losses[-1])
    model.eval()
    correct = total = 0
    with torch.no_grad():
        for img, gt in loader:
            out, _ = model(img, None)
            correct += (out[:, :, 0].argmax(1) == gt).sum().item()
            total += gt.numel()
    assert correct / total >= 0.75, correct / total
    # EM actually moved the priors off their 1/K init
    own = model.last_layer.weight.data[model.prototype_class_identity.t() == 1]
    assert (own - 1.0 / K).abs().max() > 1e-3
