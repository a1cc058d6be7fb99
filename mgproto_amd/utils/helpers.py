"""Small shared helpers (reference ``utils/helpers.py``), cv2-free.

``find_high_activation_crop`` uses scipy.ndimage connected components instead
of OpenCV (not installed here); semantics match reference helpers.py:38-74.
"""

import os
import time

import numpy as np
import torch

try:
    from scipy import ndimage
except ImportError:  # pragma: no cover
    ndimage = None


def list_of_distances(X: torch.Tensor, Y: torch.Tensor) -> torch.Tensor:
    """Pairwise squared L2 distances [n, m] (reference helpers.py:13)."""
    return torch.sum((torch.unsqueeze(X, dim=2) - torch.unsqueeze(Y.t(), dim=0)) ** 2,
                     dim=1)


def make_one_hot(target, target_one_hot):
    target = target.view(-1, 1)
    target_one_hot.zero_()
    target_one_hot.scatter_(dim=1, index=target, value=1.)


def print_and_write(s, file):
    """Reference helpers.py:33-35."""
    print(s)
    file.write(s + '\n')


def makedir(path):
    if not os.path.exists(path):
        os.makedirs(path, exist_ok=True)


def datestr():
    now = time.gmtime()
    return '{}{:02}{:02}_{:02}{:02}'.format(now.tm_year, now.tm_mon, now.tm_mday,
                                            now.tm_hour, now.tm_min)


def find_high_activation_crop(activation_map: np.ndarray, percentile: float = 95):
    """Bounding box of the connected high-activation component containing the
    activation peak (reference helpers.py:38-74, scipy instead of cv2)."""
    threshold = np.percentile(activation_map, percentile)
    mask = (activation_map >= threshold).astype(np.uint8)

    hi = np.unravel_index(np.argmax(activation_map), activation_map.shape)
    if ndimage is not None:
        labeled, n_labels = ndimage.label(mask, structure=np.ones((3, 3)))
        peak_label = labeled[hi[0], hi[1]]
        if peak_label > 0:
            mask = (labeled == peak_label).astype(np.uint8)

    ys, xs = np.where(mask > 0)
    if len(ys) == 0:
        return (hi[0], hi[0] + 1, hi[1], hi[1] + 1)
    lower_y, upper_y = int(ys.min()), int(ys.max())
    lower_x, upper_x = int(xs.min()), int(xs.max())
    return (lower_y, upper_y + 1, lower_x, upper_x + 1)


def setup_miopen_db():
    """Seed MIOpen's user find-db from the in-repo copy (gfx950).

    MIOpen's exhaustive find (torch.backends.cudnn.benchmark) costs ~2-3
    minutes per fresh box; the resulting tuning db is tiny text, so we ship
    it and point MIOPEN_USER_DB_PATH at a writable copy. No-op if the env
    var is already set or the db directory is absent.
    """
    import shutil
    import tempfile
    if os.environ.get('MIOPEN_USER_DB_PATH'):
        return
    src = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                       'miopen_db')
    if not os.path.isdir(src) or not os.listdir(src):
        return
    dst = os.path.join(tempfile.gettempdir(), 'mgproto_miopen_db')
    os.makedirs(dst, exist_ok=True)
    for f in os.listdir(src):
        tgt = os.path.join(dst, f)
        if not os.path.exists(tgt):
            # atomic publish: concurrent ranks must never observe a
            # half-copied db file
            tmp = tgt + f'.tmp{os.getpid()}'
            shutil.copy2(os.path.join(src, f), tmp)
            os.replace(tmp, tgt)
    os.environ['MIOPEN_USER_DB_PATH'] = dst
