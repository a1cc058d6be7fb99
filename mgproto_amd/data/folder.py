"""ImageFolder-style datasets (PIL-based; torchvision is not installed).

``ImageFolder`` mirrors torchvision semantics (class-per-subdirectory,
sorted class names). Items are ``(image, label, index)`` — the global index
travels with the sample so the distributed push merge has a rank-invariant
image identity (SURVEY.md hard part #4). ``MyImageFolder`` reproduces the
reference's ((img, label), (path, label)) format (utils/helpers.py:8).
"""

import os
from typing import Callable, List, Optional, Tuple

from torch.utils.data import Dataset

IMG_EXTENSIONS = ('.jpg', '.jpeg', '.png', '.ppm', '.bmp', '.pgm', '.tif',
                  '.tiff', '.webp')


def find_classes(directory: str) -> Tuple[List[str], dict]:
    classes = sorted(e.name for e in os.scandir(directory) if e.is_dir())
    if not classes:
        raise FileNotFoundError(f'no class folders in {directory}')
    return classes, {c: i for i, c in enumerate(classes)}


def make_dataset(directory: str, class_to_idx: dict) -> List[Tuple[str, int]]:
    samples = []
    for cls in sorted(class_to_idx.keys()):
        cdir = os.path.join(directory, cls)
        for root, _, fnames in sorted(os.walk(cdir, followlinks=True)):
            for fname in sorted(fnames):
                if fname.lower().endswith(IMG_EXTENSIONS):
                    samples.append((os.path.join(root, fname),
                                    class_to_idx[cls]))
    return samples


class ImageFolder(Dataset):
    def __init__(self, root: str, transform: Optional[Callable] = None,
                 decode_size: Optional[int] = None):
        self.root = root
        self.classes, self.class_to_idx = find_classes(root)
        self.samples = make_dataset(root, self.class_to_idx)
        self.imgs = self.samples  # torchvision-compat alias
        self.transform = transform
        # When set, JPEGs are decoded by libjpeg at the smallest 1/1..1/8
        # scale that keeps both dimensions >= decode_size (PIL draft()) —
        # skipping most of the IDCT work for large sources. Opt-in: it
        # slightly reduces the resolution augmentations sample from.
        self.decode_size = decode_size

    def __len__(self):
        return len(self.samples)

    def loader(self, path):
        from PIL import Image
        with open(path, 'rb') as f:
            img = Image.open(f)
            if self.decode_size and img.format == 'JPEG':
                img.draft('RGB', (self.decode_size, self.decode_size))
            return img.convert('RGB')

    def __getitem__(self, index):
        path, target = self.samples[index]
        img = self.loader(path)
        if self.transform is not None:
            img = self.transform(img)
        return img, target, index


class MyImageFolder(ImageFolder):
    """Reference-format items: ((img, label), (path, label))."""

    def __getitem__(self, index):
        path, target = self.samples[index]
        img = self.loader(path)
        if self.transform is not None:
            img = self.transform(img)
        return (img, target), (path, target)
