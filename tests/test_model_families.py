"""CPU train-step smoke for every backbone family MGProto supports
(reference models/{resnet,densenet,vgg}_features.py): construct, one
fwd+bwd with mining+enqueue, gradient reaches the backbone, checkpoint
round-trips. The flagship R50 paths get deeper coverage elsewhere."""

import pytest
import torch
import torch.nn.functional as F

from mgproto_amd.model import construct_MGProto

ARCHS = ['resnet18', 'densenet121', 'densenet161', 'vgg11', 'vgg19',
         'vgg16_bn']


@pytest.mark.parametrize('arch', ARCHS)
def test_family_train_step_and_checkpoint(arch, tmp_path):
    torch.manual_seed(0)
    C, K, d = 4, 2, 16
    model = construct_MGProto(arch, pretrained=False, img_size=64,
                              prototype_shape=(C * K, d, 1, 1),
                              num_classes=C, add_on_layers_type='regular',
                              sz_embedding=8, mem_capacity=8, mine_K=2)
    model.train()
    x = torch.randn(4, 3, 64, 64)
    y = torch.tensor([0, 1, 2, 3])
    out, aux = model(x, y)
    assert out.shape[0] == 4 and out.shape[1] == C
    loss = F.cross_entropy(out[:, :, 0], y) + 0.1 * aux.square().mean()
    loss.backward()
    gnorm = sum(p.grad.abs().sum() for p in model.features.parameters()
                if p.grad is not None)
    assert torch.isfinite(loss) and gnorm > 0
    # enqueue happened for the GT classes
    assert int(model.queue.mem_len.sum()) > 0

    p = tmp_path / 'm.pth'
    torch.save(model.state_dict(), p)
    m2 = construct_MGProto(arch, pretrained=False, img_size=64,
                           prototype_shape=(C * K, d, 1, 1), num_classes=C,
                           add_on_layers_type='regular', sz_embedding=8,
                           mem_capacity=8, mine_K=2)
    m2.load_state_dict(torch.load(p, weights_only=True))
    m2.eval()
    with torch.no_grad():
        out2, _ = m2(x, None)
    assert out2.shape == out.shape


def test_bottleneck_addon_forward():
    """The reference's default add-on ('bottleneck': halving 1x1 conv chain
    ending in Sigmoid, reference model.py:106-124) builds and trains."""
    torch.manual_seed(0)
    C, K, d = 4, 2, 16
    model = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(C * K, d, 1, 1),
                              num_classes=C, add_on_layers_type='bottleneck',
                              sz_embedding=8, mem_capacity=8, mine_K=2)
    # chain halves 512 -> ... -> d with a final Sigmoid
    import torch.nn as nn
    assert isinstance(list(model.add_on_layers)[-1], nn.Sigmoid)
    x = torch.randn(2, 3, 64, 64)
    y = torch.tensor([0, 1])
    out, _ = model(x, y)
    F.cross_entropy(out[:, :, 0], y).backward()
    assert out.shape[:2] == (2, C)


def test_vgg_vanilla_classifier():
    """VGG_vanilla (plain classifier head, reference vgg_features.py:110)."""
    from mgproto_amd.models.vgg import VGG_vanilla
    m = VGG_vanilla(num_classes=5, pretrained=False)
    out = m(torch.randn(1, 3, 224, 224))
    assert out.shape == (1, 5)
