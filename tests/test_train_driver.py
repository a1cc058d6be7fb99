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


def test_train_driver_two_ranks(tmp_path):
    """Full train.py run on 2 gloo ranks (tiny synthetic config, one epoch
    including a push epoch): exercises make_dp_correct + grad reducer +
    distributed push through the real driver wiring."""
    import socket
    with socket.socket() as s:
        s.bind(('127.0.0.1', 0))
        port = s.getsockname()[1]
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update({'MGPROTO_TINY_TEST': '1', 'RANK': str(rank),
                    'LOCAL_RANK': str(rank), 'WORLD_SIZE': '2',
                    'MASTER_ADDR': '127.0.0.1', 'MASTER_PORT': str(port)})
        cmd = [sys.executable, os.path.join(ROOT, 'train.py'),
               '-arch', 'resnet18', '-mem_sz', '8', '-mine_level', '3',
               '-aux_emb_sz', '8', '--epochs', '1',
               '--out', str(tmp_path / 'run2')]
        procs.append(subprocess.Popen(cmd, stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, text=True,
                                      env=env, cwd=ROOT))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=900)
        outs.append(out)
    for p, out in zip(procs, outs):
        assert p.returncode == 0, out[-3000:]
    out_dir = tmp_path / 'run2'
    assert (out_dir / 'latest.pth').is_file()
    recs = [json.loads(l) for l in open(out_dir / 'metrics.jsonl')]
    assert any('test/acc' in r for r in recs)


def test_auto_resume_elastic_restart(tmp_path):
    """A relaunch with the same --out continues from the newest checkpoint
    (elastic-restart semantics); --no-auto-resume starts fresh."""
    _run_train(tmp_path)                                   # epoch 0
    r = _run_train(tmp_path, extra=('--epochs', '2'))      # relaunch
    log_text = open(tmp_path / 'run' / 'train.log').read()
    assert 'resumed from' in log_text
    state = torch.load(tmp_path / 'run' / 'latest.pth', map_location='cpu',
                       weights_only=False)
    assert state['epoch'] == 1                             # continued, not redone

    r = _run_train(tmp_path, extra=('--epochs', '1', '--no-auto-resume'))
    state = torch.load(tmp_path / 'run' / 'latest.pth', map_location='cpu',
                       weights_only=False)
    assert state['epoch'] == 0                             # fresh run


def test_train_driver_seed_determinism(tmp_path):
    """Two identical seeded runs end with bit-identical model state."""
    import hashlib

    def digest(p):
        sd = torch.load(p, map_location='cpu', weights_only=False)['model']
        h = hashlib.sha256()
        for k in sorted(sd):
            h.update(k.encode())
            h.update(sd[k].float().contiguous().numpy().tobytes())
        return h.hexdigest()

    _run_train(tmp_path / 'a', extra=('--seed', '7', '--no-auto-resume'))
    _run_train(tmp_path / 'b', extra=('--seed', '7', '--no-auto-resume'))
    assert digest(tmp_path / 'a' / 'run' / 'latest.pth') \
        == digest(tmp_path / 'b' / 'run' / 'latest.pth')


def test_train_driver_regular_upsample(tmp_path):
    """Driver run with the flagship add-on (convs-then-upsample, 2x grid)."""
    _run_train(tmp_path, extra=('--addon', 'regular_upsample',))
    assert (tmp_path / 'run' / 'latest.pth').is_file()


def test_train_driver_ood_eval(tmp_path):
    """--ood-eval evaluates FPR95/AUROC against the synthetic OoD sets."""
    _run_train(tmp_path, extra=('--ood-eval',))
    log_text = open(tmp_path / 'run' / 'train.log').read()
    assert 'FPR95_1' in log_text and 'FPR95_2' in log_text


def test_summarize_metrics_tool(tmp_path):
    _run_train(tmp_path)
    r = subprocess.run([sys.executable, 'tools/summarize_metrics.py',
                        str(tmp_path / 'run' / 'metrics.jsonl')],
                       capture_output=True, text=True, timeout=120, cwd=ROOT)
    assert r.returncode == 0, r.stderr
    assert 'best test acc' in r.stdout and 'epoch' in r.stdout
