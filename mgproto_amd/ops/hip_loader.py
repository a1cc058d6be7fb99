"""Loader for the in-tree mgproto_hip extension (.so built for gfx950).

The extension is built IN-TREE (mgproto_amd/ops/_mgproto_hip.so) by
``mgproto_amd.ops.build`` / ``__graft_entry__.build()`` so the binary travels
with repo snapshots; a torch-extensions JIT cache would not.
"""

import importlib.util
import os

_HERE = os.path.dirname(os.path.abspath(__file__))
SO_PATH = os.path.join(_HERE, '_mgproto_hip.so')
CPU_SO_PATH = os.path.join(_HERE, '_mgproto_cpu.so')

_mod = None
_cpu_mod = None


def load():
    global _mod
    if _mod is not None:
        return _mod
    if not os.path.isfile(SO_PATH):
        raise FileNotFoundError(f'{SO_PATH} not built')
    import torch  # noqa: F401  (extension links against torch libs)
    spec = importlib.util.spec_from_file_location('_mgproto_hip', SO_PATH)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    _mod = mod
    return _mod


def load_cpu():
    """Native CPU augmentation core (optional; python fallback exists)."""
    global _cpu_mod
    if _cpu_mod is not None:
        return _cpu_mod
    if not os.path.isfile(CPU_SO_PATH):
        raise FileNotFoundError(f'{CPU_SO_PATH} not built')
    import torch  # noqa: F401
    spec = importlib.util.spec_from_file_location('_mgproto_cpu', CPU_SO_PATH)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    _cpu_mod = mod
    return _cpu_mod
