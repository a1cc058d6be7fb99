"""CPU-testable GraphedStep logic: signature tracking, re-capture resets,
batch-size gating, broken-capture fallback. The capture/replay itself is
GPU-only (tests/test_graph_train_gpu.py)."""

import torch

from mgproto_amd.engine.graphstep import GraphedStep, graphs_enabled


class _Net(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.a = torch.nn.Linear(4, 4)
        self.b = torch.nn.Linear(4, 2)

    def forward(self, x, gt):
        return self.b(self.a(x)), x


def _gs(bs=4):
    net = _Net()
    return GraphedStep(net, None, {'crs_ent': 1.0, 'mine': 0.2, 'aux': 0.5},
                       torch.device('cpu'), bs, 8, channels_last=False), net


def test_graphs_enabled_gating(monkeypatch):
    assert not graphs_enabled(torch.device('cpu'))
    assert not graphs_enabled(None)
    # the env kill-switch wins even for cuda devices
    monkeypatch.setenv('MGPROTO_NO_GRAPH', '1')
    assert not graphs_enabled(torch.device('cuda'))


def test_matches_gates_on_batch_size():
    gs, net = _gs(bs=4)
    opt = torch.optim.SGD(net.parameters(), lr=0.1)
    full = torch.zeros(4, 3, 8, 8)
    rem = torch.zeros(3, 3, 8, 8)        # remainder batch
    assert gs.matches(full, opt, True, True)
    assert not gs.matches(rem, opt, True, True)


def test_signature_change_resets_capture_state():
    gs, net = _gs()
    opt1 = torch.optim.SGD(net.parameters(), lr=0.1)
    opt2 = torch.optim.SGD(net.parameters(), lr=0.2)
    x = torch.zeros(4, 3, 8, 8)
    assert gs.matches(x, opt1, True, True)
    gs._warm_count = 3
    gs.graph = object()                  # pretend captured
    # same signature: state kept
    assert gs.matches(x, opt1, True, True)
    assert gs.graph is not None and gs._warm_count == 3
    # optimizer swap (warm<->joint): full reset
    assert gs.matches(x, opt2, True, True)
    assert gs.graph is None and gs._warm_count == 0
    # mining / EM flips also reset
    gs.graph = object()
    assert gs.matches(x, opt2, False, True)
    assert gs.graph is None
    gs.graph = object()
    assert gs.matches(x, opt2, False, False)
    assert gs.graph is None
    # requires_grad flip (phase change on the module) resets
    gs.graph = object()
    for p in net.a.parameters():
        p.requires_grad = False
    assert gs.matches(x, opt2, False, False)
    assert gs.graph is None


def test_broken_capture_disables_graph_path():
    gs, net = _gs()
    opt = torch.optim.SGD(net.parameters(), lr=0.1)
    gs.broken = True
    assert not gs.matches(torch.zeros(4, 3, 8, 8), opt, True, True)
