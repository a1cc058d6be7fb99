"""Image transforms (PIL + torch; torchvision is not installed here).

Implements the transform set the reference training pipeline uses
(reference main.py:96-163): Resize, CenterCrop, RandomResizedCrop,
RandomHorizontalFlip, ColorJitter, RandomAffine, RandomPerspective,
ToTensor, Normalize, Compose — with torchvision-compatible semantics for
the parameters the reference passes.
"""

import math
import random
from typing import Sequence, Tuple

import numpy as np
import torch

try:
    from PIL import Image, ImageEnhance
except ImportError:  # pragma: no cover
    Image = None


class Compose:
    def __init__(self, transforms):
        self.transforms = transforms

    def __call__(self, img):
        for t in self.transforms:
            img = t(img)
        return img


class ToTensor:
    def __call__(self, img):
        if isinstance(img, torch.Tensor):
            return img
        arr = np.asarray(img, dtype=np.uint8)
        if arr.ndim == 2:
            arr = arr[:, :, None]
        t = torch.from_numpy(arr.copy()).permute(2, 0, 1).float() / 255.0
        return t


class Normalize:
    def __init__(self, mean, std):
        self.mean = torch.tensor(mean).view(-1, 1, 1)
        self.std = torch.tensor(std).view(-1, 1, 1)

    def __call__(self, t):
        return (t - self.mean) / self.std


class Resize:
    """int -> resize shorter side; (h, w) -> exact size (PIL bilinear)."""

    def __init__(self, size):
        self.size = size

    def __call__(self, img):
        if isinstance(self.size, int):
            w, h = img.size
            if w <= h:
                nw, nh = self.size, int(round(h * self.size / w))
            else:
                nh, nw = self.size, int(round(w * self.size / h))
        else:
            nh, nw = self.size
        return img.resize((nw, nh), Image.BILINEAR)


class CenterCrop:
    def __init__(self, size):
        self.size = (size, size) if isinstance(size, int) else size

    def __call__(self, img):
        w, h = img.size
        th, tw = self.size
        left = int(round((w - tw) / 2.0))
        top = int(round((h - th) / 2.0))
        return img.crop((left, top, left + tw, top + th))


class RandomHorizontalFlip:
    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, img):
        if random.random() < self.p:
            return img.transpose(Image.FLIP_LEFT_RIGHT)
        return img


class RandomResizedCrop:
    def __init__(self, size, scale=(0.08, 1.0), ratio=(3. / 4., 4. / 3.)):
        self.size = (size, size) if isinstance(size, int) else tuple(size)
        self.scale = scale
        self.ratio = ratio

    def __call__(self, img):
        w, h = img.size
        area = w * h
        for _ in range(10):
            target_area = random.uniform(*self.scale) * area
            log_ratio = (math.log(self.ratio[0]), math.log(self.ratio[1]))
            aspect = math.exp(random.uniform(*log_ratio))
            cw = int(round(math.sqrt(target_area * aspect)))
            ch = int(round(math.sqrt(target_area / aspect)))
            if 0 < cw <= w and 0 < ch <= h:
                left = random.randint(0, w - cw)
                top = random.randint(0, h - ch)
                crop = img.crop((left, top, left + cw, top + ch))
                return crop.resize((self.size[1], self.size[0]), Image.BILINEAR)
        # fallback: center crop
        return CenterCrop(min(w, h))(img).resize(
            (self.size[1], self.size[0]), Image.BILINEAR)


class ColorJitter:
    """brightness/contrast/saturation ranges as (lo, hi); hue as (lo, hi)."""

    def __init__(self, brightness=None, contrast=None, saturation=None, hue=None):
        self.brightness = self._pair(brightness)
        self.contrast = self._pair(contrast)
        self.saturation = self._pair(saturation)
        self.hue = hue if (hue is None or isinstance(hue, (tuple, list))) \
            else (-hue, hue)

    @staticmethod
    def _pair(v):
        if v is None:
            return None
        if isinstance(v, (tuple, list)):
            return tuple(v)
        return (max(0.0, 1 - v), 1 + v)

    def __call__(self, img):
        ops = []
        if self.brightness:
            f = random.uniform(*self.brightness)
            ops.append(lambda im: ImageEnhance.Brightness(im).enhance(f))
        if self.contrast:
            f = random.uniform(*self.contrast)
            ops.append(lambda im: ImageEnhance.Contrast(im).enhance(f))
        if self.saturation:
            f = random.uniform(*self.saturation)
            ops.append(lambda im: ImageEnhance.Color(im).enhance(f))
        if self.hue:
            shift = random.uniform(*self.hue)
            ops.append(lambda im: self._hue(im, shift))
        random.shuffle(ops)
        for op in ops:
            img = op(img)
        return img

    @staticmethod
    def _hue(img, shift):
        hsv = img.convert('HSV')
        arr = np.asarray(hsv, dtype=np.uint8).copy()
        arr[:, :, 0] = (arr[:, :, 0].astype(np.int16)
                        + int(shift * 255)) % 256
        return Image.fromarray(arr, 'HSV').convert('RGB')


class RandomAffine:
    def __init__(self, degrees=0, translate=None, scale=None, shear=None):
        self.degrees = (-degrees, degrees) if isinstance(degrees, (int, float)) \
            else tuple(degrees)
        self.translate = translate
        self.scale = scale
        if shear is None:
            self.shear = None
        elif isinstance(shear, (int, float)):
            self.shear = (-shear, shear)
        else:
            self.shear = tuple(shear)

    def __call__(self, img):
        w, h = img.size
        angle = math.radians(random.uniform(*self.degrees))
        tx = ty = 0.0
        if self.translate is not None:
            tx = random.uniform(-self.translate[0], self.translate[0]) * w
            ty = random.uniform(-self.translate[1], self.translate[1]) * h
        s = random.uniform(*self.scale) if self.scale else 1.0
        shear_x = math.radians(random.uniform(*self.shear)) if self.shear else 0.0

        # inverse affine matrix around the image center (torchvision semantics)
        cx, cy = w * 0.5, h * 0.5
        cos_a, sin_a = math.cos(angle), math.sin(angle)
        # forward: R(angle) * Shear(shear_x) * s, then translate
        a = s * cos_a
        b = s * (-sin_a + cos_a * math.tan(shear_x))
        c_ = s * sin_a
        d = s * (cos_a + sin_a * math.tan(shear_x))
        det = a * d - b * c_
        if abs(det) < 1e-8:
            return img
        ia, ib = d / det, -b / det
        ic, id_ = -c_ / det, a / det
        # x_src = ia*(x - cx - tx) + ib*(y - cy - ty) + cx, similarly y_src
        coeffs = (ia, ib, cx - ia * (cx + tx) - ib * (cy + ty),
                  ic, id_, cy - ic * (cx + tx) - id_ * (cy + ty))
        return img.transform((w, h), Image.AFFINE, coeffs, Image.BILINEAR)


class RandomPerspective:
    def __init__(self, distortion_scale=0.5, p=0.5):
        self.distortion_scale = distortion_scale
        self.p = p

    def __call__(self, img):
        if random.random() >= self.p:
            return img
        w, h = img.size
        d = self.distortion_scale
        dx, dy = int(d * w / 2), int(d * h / 2)
        tl = (random.randint(0, dx), random.randint(0, dy))
        tr = (w - 1 - random.randint(0, dx), random.randint(0, dy))
        br = (w - 1 - random.randint(0, dx), h - 1 - random.randint(0, dy))
        bl = (random.randint(0, dx), h - 1 - random.randint(0, dy))
        start = [(0, 0), (w - 1, 0), (w - 1, h - 1), (0, h - 1)]
        end = [tl, tr, br, bl]
        coeffs = _perspective_coeffs(end, start)
        return img.transform((w, h), Image.PERSPECTIVE, coeffs, Image.BILINEAR)


def _perspective_coeffs(src: Sequence[Tuple[int, int]],
                        dst: Sequence[Tuple[int, int]]):
    """Solve the 8-dof homography mapping dst -> src (PIL convention)."""
    A = []
    B = []
    for (x, y), (u, v) in zip(dst, src):
        A.append([x, y, 1, 0, 0, 0, -u * x, -u * y])
        A.append([0, 0, 0, x, y, 1, -v * x, -v * y])
        B.extend([u, v])
    A = np.array(A, dtype=np.float64)
    B = np.array(B, dtype=np.float64)
    res = np.linalg.lstsq(A, B, rcond=None)[0]
    return tuple(res.tolist())
