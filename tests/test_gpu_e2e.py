"""GPU end-to-end: flagship config forward+backward+EM+push on one MI355X."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu


def test_flagship_step_and_push():
    from mgproto_amd.model import construct_MGProto
    from mgproto_amd import ops
    assert ops.native_available(), 'HIP extension must be loaded on GPU'

    dev = torch.device('cuda')
    torch.manual_seed(0)
    C, K, d = 200, 10, 64
    model = construct_MGProto('resnet50', pretrained=False, img_size=224,
                              prototype_shape=(C * K, d, 1, 1), num_classes=C,
                              add_on_layers_type='regular_upsample',
                              sz_embedding=32, mem_capacity=800,
                              mine_K=20).to(dev)
    model.features = model.features.to(memory_format=torch.channels_last)
    opt = torch.optim.Adam(model.parameters(), lr=1e-4)
    model.train()

    x = torch.randn(8, 3, 224, 224, device=dev)
    gt = torch.randint(0, C, (8,), device=dev)
    with torch.autocast(device_type='cuda', dtype=torch.bfloat16):
        logits, emb = model(x, gt)
    assert logits.shape == (8, C, 20)
    loss = F.cross_entropy(logits.float()[:, :, 0], gt)
    loss.backward()
    opt.step()
    assert torch.isfinite(loss)
    assert int(model.queue.mem_len.sum()) > 0

    # fill queue, EM on device
    with torch.no_grad():
        mem = F.normalize(torch.randn(C * 800, d, device=dev), dim=1)
        labels = torch.arange(C, device=dev).repeat_interleave(800)
        model.queue.push(mem, labels)
        model.memory_updated_cls[:] = True
    means_before = model.prototype_means.data.clone()
    model.update_GMM()
    torch.cuda.synchronize()
    assert not torch.allclose(model.prototype_means.data, means_before)
    assert torch.isfinite(model.prototype_means.data).all()

    # push_forward shapes + finiteness
    model.eval()
    with torch.no_grad():
        bf, dist = model.push_forward(x[:2])
    assert dist.shape == (2, C * K, 28, 28)
    assert torch.isfinite(dist).all()


def test_gpu_matches_cpu_model_outputs():
    """Same weights, same input: GPU (HIP kernels) vs CPU (torch reference)."""
    from mgproto_amd.model import construct_MGProto
    torch.manual_seed(0)
    C, K, d = 20, 5, 64
    model = construct_MGProto('resnet18', pretrained=False, img_size=64,
                              prototype_shape=(C * K, d, 1, 1), num_classes=C,
                              add_on_layers_type='regular', sz_embedding=16,
                              mem_capacity=8, mine_K=4)
    model.eval()
    x = torch.randn(4, 3, 64, 64)
    with torch.no_grad():
        out_cpu, emb_cpu = model(x, None)
    gpu = model.to('cuda')
    with torch.no_grad():
        out_gpu, emb_gpu = gpu(x.cuda(), None)
    # fp32 end to end on both paths; conv nondeterminism tolerance
    assert torch.allclose(out_gpu.cpu(), out_cpu, atol=1e-3, rtol=1e-3)
    assert torch.allclose(emb_gpu.cpu(), emb_cpu, atol=1e-4, rtol=1e-4)


@pytest.mark.gpu
def test_device_prefetcher_stream_path():
    """Prefetcher on CUDA: same batches, device tensors, overlap-safe."""
    from torch.utils.data import DataLoader

    from mgproto_amd.data.prefetch import DevicePrefetcher
    from mgproto_amd.data.synthetic import SyntheticImages

    ds = SyntheticImages(n=12, num_classes=3, img_size=32)
    loader = DataLoader(ds, batch_size=4, pin_memory=True)
    pf = DevicePrefetcher(loader, torch.device('cuda', 0))
    got = list(pf)
    ref = list(DataLoader(ds, batch_size=4))
    assert len(got) == 3
    for (a, b, c), (x, y, z) in zip(got, ref):
        assert a.is_cuda and b.is_cuda
        assert torch.equal(a.cpu(), x) and torch.equal(b.cpu(), y)
        # consume with a compute op to exercise the event ordering
        _ = (a * 2).sum().item()
