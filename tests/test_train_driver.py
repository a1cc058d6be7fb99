"""train.py driver: tiny synthetic run end-to-end on CPU, plus resume."""

import json
import os
import subprocess
import sys

import torch

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_train(tmp_path, extra=()):
    env = dict(os.environ)
    env['MGPROTO_TINY_TEST'] = '1'
    cmd = [sys.executable, os.path.join(ROOT, 'train.py'),
           '-arch', 'resnet18', '-mem_sz', '8', '-mine_level', '3',
           '-aux_emb_sz', '8',
           '--epochs', '1', '--out', str(tmp_path / 'run'), *extra]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=900,
                       env=env, cwd=ROOT)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    return r


def test_train_one_epoch_and_resume(tmp_path):
    _run_train(tmp_path)
    out = tmp_path / 'run'
    assert (out / 'train.log').is_file()
    assert (out / 'latest.pth').is_file()
    assert (out / 'metrics.jsonl').is_file()
    recs = [json.loads(l) for l in open(out / 'metrics.jsonl')]
    assert any('test/acc' in r for r in recs)

    # resume from the checkpoint
    state = torch.load(out / 'latest.pth', map_location='cpu',
                       weights_only=False)
    assert state['epoch'] == 0
    assert 'model' in state and 'optimizers' in state
    # checkpoint carries the reference layout
    assert 'queue.cls0' in state['model']
    assert 'prototype_means' in state['model']
