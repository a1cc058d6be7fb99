"""Every module in the package imports cleanly (catches import-time
regressions in rarely-exercised corners)."""

import importlib
import os
import pkgutil

import mgproto_amd


def test_all_modules_import():
    pkg_dir = os.path.dirname(mgproto_amd.__file__)
    failures = []
    for mod in pkgutil.walk_packages([pkg_dir], prefix='mgproto_amd.'):
        try:
            importlib.import_module(mod.name)
        except Exception as e:  # noqa: BLE001
            failures.append((mod.name, repr(e)))
    assert not failures, failures


def test_top_level_drivers_import():
    import bench       # noqa: F401
    import serve       # noqa: F401
    import train       # noqa: F401
    import eval_consistency   # noqa: F401
    import eval_purity        # noqa: F401
    import eval_stability     # noqa: F401
    import __graft_entry__    # noqa: F401


def test_check_env_tool_runs():
    import os
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run([sys.executable, 'tools/check_env.py'],
                       capture_output=True, text=True, timeout=120, cwd=root)
    # exit 1 here (no GPU in CI) but every non-GPU check must PASS
    assert 'torch' in r.stdout
    for line in r.stdout.splitlines():
        if line.startswith('FAIL'):
            assert 'GPU' in line or 'gfx950' in line or 'loads' in line, line
