"""Serving path: InferenceEngine semantics + the FastAPI surface."""

import io

import numpy as np
import pytest
import torch
import torch.nn.functional as F

from mgproto_amd.model import construct_MGProto
from mgproto_amd.serving import InferenceEngine, create_app


@pytest.fixture(scope='module')
def engine():
    torch.manual_seed(0)
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(20, 16, 1, 1), num_classes=5,
                          add_on_layers_type='regular', sz_embedding=8,
                          mem_capacity=4, mine_K=3)
    return InferenceEngine(m, torch.device('cpu'), amp=False)


def test_predict_schema(engine):
    x = torch.randn(2, 3, 64, 64)
    res = engine.predict(x, topk_classes=3, explain_topk=2)
    assert len(res) == 2
    for r in res:
        assert 0 <= r['pred_class'] < 5
        assert len(r['top_classes']) == 3
        probs = [t['prob'] for t in r['top_classes']]
        assert probs == sorted(probs, reverse=True)
        assert r['density_pX'] > 0
        assert len(r['explanations']) == 2
        for e in r['explanations']:
            # explanations come from the predicted class's prototypes
            assert e['prototype'] // 4 == r['pred_class']
            assert 'rf_bbox_yxyx' in e and len(e['rf_bbox_yxyx']) == 4
            y1, y2, x1, x2 = e['rf_bbox_yxyx']
            assert 0 <= y1 <= y2 <= 64 and 0 <= x1 <= x2 <= 64


def test_predict_consistent_with_forward(engine):
    """Serving logits must equal model.forward's level-0 output (eval)."""
    torch.manual_seed(1)
    x = torch.randn(2, 3, 64, 64)
    res = engine.predict(x, topk_classes=5)
    with torch.no_grad():
        logits, _ = engine.model(x, None)
    want = F.softmax(logits[:, :, 0], dim=1)
    for b in range(2):
        got = sorted(res[b]['top_classes'], key=lambda t: t['class'])
        for t in got:
            assert abs(t['prob'] - float(want[b, t['class']])) < 1e-4


def test_fastapi_endpoints(engine):
    pytest.importorskip('fastapi')
    from fastapi.testclient import TestClient
    from PIL import Image

    app = create_app(engine, class_names=[f'bird_{i}' for i in range(5)])
    client = TestClient(app)

    r = client.get('/healthz')
    assert r.status_code == 200 and r.json()['status'] == 'ok'

    r = client.get('/model_info')
    assert r.json()['num_classes'] == 5

    rng = np.random.RandomState(0)
    buf = io.BytesIO()
    Image.fromarray(rng.randint(0, 255, (80, 90, 3), dtype=np.uint8)) \
        .save(buf, format='PNG')
    r = client.post('/predict', content=buf.getvalue(),
                    headers={'content-type': 'image/png'})
    assert r.status_code == 200, r.text
    body = r.json()
    assert 'pred_class' in body and 'explanations' in body
    assert body['top_classes'][0]['name'].startswith('bird_')


@pytest.mark.gpu
def test_serving_graph_capture_gpu():
    """Graph-captured serving forward == eager forward on GPU."""
    torch.manual_seed(0)
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(20, 16, 1, 1), num_classes=5,
                          add_on_layers_type='regular', sz_embedding=8,
                          mem_capacity=4, mine_K=3).cuda()
    m.features = m.features.to(memory_format=torch.channels_last)
    eng = InferenceEngine(m)
    x = torch.randn(4, 3, 64, 64)
    want = eng.predict(x, topk_classes=3)
    eng.capture(4, 64)
    got = eng.predict(x, topk_classes=3)
    for w, g in zip(want, got):
        assert w['pred_class'] == g['pred_class']
        assert abs(w['density_pX'] - g['density_pX']) < 1e-3 * (1 + abs(w['density_pX']))


def test_serve_cli_loads_any_config_checkpoint(tmp_path):
    """serve.py infers mem/embedding sizes from the checkpoint (same as the
    eval drivers) — loading a non-default-config checkpoint must not crash.
    Exercised via the loader path, not the HTTP server."""
    import os
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    from mgproto_amd.model import construct_MGProto
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(6, 16, 1, 1), num_classes=2,
                          add_on_layers_type='regular', sz_embedding=8,
                          mem_capacity=8, mine_K=2)
    ckpt = tmp_path / 'm.pth'
    torch.save(m.state_dict(), str(ckpt))
    code = (
        "import sys; sys.argv = ['serve.py', '--resume', %r, '--arch',"
        " 'resnet18', '--classes', '2', '--proto-per-class', '3',"
        " '--proto-dim', '16', '--img', '64', '--addon', 'regular'];"
        "import unittest.mock as mock, uvicorn;"
        "mock.patch.object(uvicorn, 'run').start();"
        "import serve; serve.main(); print('LOADED-OK')" % str(ckpt))
    r = subprocess.run([sys.executable, '-c', code], capture_output=True,
                       text=True, timeout=300, cwd=root)
    assert r.returncode == 0 and 'LOADED-OK' in r.stdout, \
        r.stdout[-2000:] + r.stderr[-2000:]


def test_serve_bench_tool_cpu(tmp_path):
    import json
    import os
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, 'tools/serve_bench.py', '--arch', 'resnet18',
         '--classes', '4', '--proto-per-class', '2', '--proto-dim', '16',
         '--img', '64', '--batch', '2', '--iters', '3', '--warmup', '1'],
        capture_output=True, text=True, timeout=600, cwd=root)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    rec = json.loads([l for l in r.stdout.splitlines()
                      if l.startswith('{')][-1])
    assert rec['mode'] == 'eager' and rec['images_per_sec'] > 0
