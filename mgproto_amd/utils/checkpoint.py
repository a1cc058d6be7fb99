"""Checkpoint save/load.

Reference behaviour (``utils/save.py:5-12``): save the full ``state_dict``
to ``{model_dir}/{name}{accuracy:.4f}.pth`` when accuracy exceeds a target.
Extended here with a full resume path (model + optimizers + schedulers +
epoch + RNG), which the reference lacks (SURVEY.md §5 checkpoint/resume).
"""

import os
from typing import Dict, Optional

import torch


def save_model_w_condition(model, model_dir, model_name, accu, target_accu,
                           log=print):
    if accu > target_accu:
        log('\tabove {0:.2f}%'.format(target_accu * 100))
        torch.save(obj=model.state_dict(),
                   f=os.path.join(model_dir, (model_name + '{0:.4f}.pth').format(accu)))


def infer_ctor_kwargs_from_state(sd: dict) -> dict:
    """Shape-bearing ``construct_MGProto`` kwargs that the reference CLI
    does not expose, read off a checkpoint's state_dict so any training
    config loads (memory capacity per class, aux embedding size)."""
    kw = {}
    if 'queue.cls0' in sd:
        kw['mem_capacity'] = sd['queue.cls0'].shape[0]
    if 'embedding.weight' in sd:
        kw['sz_embedding'] = sd['embedding.weight'].shape[0]
    return kw


def save_train_state(path: str, model, optimizers: Dict[str, object],
                     schedulers: Dict[str, object], epoch: int,
                     extra: Optional[dict] = None):
    import random

    import numpy as np
    state = {
        'model': model.state_dict(),
        # EM Adam moments live outside the (reference-layout) state_dict
        'em_state': (model.em_state_dict()
                     if hasattr(model, 'em_state_dict') else None),
        'optimizers': {k: v.state_dict() for k, v in optimizers.items() if v is not None},
        'schedulers': {k: v.state_dict() for k, v in schedulers.items() if v is not None},
        'epoch': epoch,
        'torch_rng': torch.get_rng_state(),
        'cuda_rng': (torch.cuda.get_rng_state_all()
                     if torch.cuda.is_available() else None),
        'py_rng': random.getstate(),
        'np_rng': np.random.get_state(),
        'extra': extra or {},
    }
    tmp = path + '.tmp'
    torch.save(state, tmp)
    os.replace(tmp, path)


def load_train_state(path: str, model, optimizers: Dict[str, object] = None,
                     schedulers: Dict[str, object] = None,
                     map_location='cpu') -> dict:
    state = torch.load(path, map_location=map_location, weights_only=False)
    model.load_state_dict(state['model'])
    if state.get('em_state') is not None and hasattr(model, 'load_em_state'):
        model.load_em_state(state['em_state'])
    for k, v in (optimizers or {}).items():
        if v is not None and k in state.get('optimizers', {}):
            v.load_state_dict(state['optimizers'][k])
    for k, v in (schedulers or {}).items():
        if v is not None and k in state.get('schedulers', {}):
            v.load_state_dict(state['schedulers'][k])
    if state.get('torch_rng') is not None:
        torch.set_rng_state(state['torch_rng'].cpu().to(torch.uint8))
    cuda_rng = state.get('cuda_rng')
    if (cuda_rng is not None and torch.cuda.is_available()
            and len(cuda_rng) == torch.cuda.device_count()):
        torch.cuda.set_rng_state_all([t.cpu().to(torch.uint8)
                                      for t in cuda_rng])
    if state.get('py_rng') is not None:
        import random
        random.setstate(state['py_rng'])
    if state.get('np_rng') is not None:
        import numpy as np
        np.random.set_state(state['np_rng'])
    return state
