"""CUB-200-2011 evaluation dataset (reference ``utils/datasets.py``).

Returns (image, target, img_id) from the CUB metadata CSVs. PIL loader
(torchvision's default_loader is unavailable here).
"""

import os

import pandas as pd
from torch.utils.data import Dataset


def pil_loader(path):
    from PIL import Image
    with open(path, 'rb') as f:
        return Image.open(f).convert('RGB')


class Cub2011Eval(Dataset):
    base_folder = 'images'

    def __init__(self, root, train=True, transform=None):
        self.root = os.path.expanduser(root)
        self.transform = transform
        self.loader = pil_loader
        self.train = train
        if not self._check_integrity():
            raise RuntimeError('Dataset not found or corrupted.')

    def _load_metadata(self):
        images = pd.read_csv(os.path.join(self.root, 'images.txt'), sep=' ',
                             names=['img_id', 'filepath'])
        labels = pd.read_csv(os.path.join(self.root, 'image_class_labels.txt'),
                             sep=' ', names=['img_id', 'target'])
        split = pd.read_csv(os.path.join(self.root, 'train_test_split.txt'),
                            sep=' ', names=['img_id', 'is_training_img'])
        data = images.merge(labels, on='img_id').merge(split, on='img_id')
        self.data = data[data.is_training_img == (1 if self.train else 0)]

    def _check_integrity(self):
        try:
            self._load_metadata()
        except Exception:  # noqa: BLE001
            return False
        for _, row in self.data.iterrows():
            if not os.path.isfile(os.path.join(self.root, self.base_folder,
                                               row.filepath)):
                return False
        return True

    def __len__(self):
        return len(self.data)

    def __getitem__(self, idx):
        sample = self.data.iloc[idx]
        path = os.path.join(self.root, self.base_folder, sample.filepath)
        target = sample.target - 1
        img = self.loader(path)
        if self.transform is not None:
            img = self.transform(img)
        return img, target, sample.img_id
