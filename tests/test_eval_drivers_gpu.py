"""GPU runs of the interpretability eval drivers (VERDICT round-1 weak #6:
they were CPU-tested only). Same mini-CUB tree and CLIs as
test_eval_drivers.py; the drivers pick cuda automatically, so on a GPU box
these exercise push_forward + the metric stacks on device."""

import pytest
import torch

import test_eval_drivers as T

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip('needs a GPU', allow_module_level=True)

mini_cub = T.mini_cub  # re-register the module-scoped fixture


def test_eval_consistency_driver_gpu(mini_cub):
    out = T._run('eval_consistency.py', mini_cub, ('--half_size', '18'))
    assert 'Consistency Score' in out


def test_eval_stability_driver_gpu(mini_cub):
    out = T._run('eval_stability.py', mini_cub, ('--half_size', '18'))
    assert 'Stability Score' in out


def test_eval_purity_drivers_gpu(mini_cub, tmp_path):
    out = T._run('eval_purity.py', mini_cub, ('--topK', '2'))
    assert 'Purity Score' in out
    out = T._run('eval_purity.py', mini_cub,
                 ('--topK', '2', '--csv', '--log_dir', str(tmp_path / 'lg')))
    assert 'CSV Purity' in out
