"""CUB-200-2011 part/bbox annotation loading.

Reference ``utils/local_parts.py`` parses the annotation txts into module
globals AT IMPORT TIME with a hardcoded path (local_parts.py:14) — here the
same parsing lives in ``CubPartAnnotations`` so the data root is an
argument and nothing happens at import. The attribute names match the
reference's globals (``id_to_path``, ``id_to_bbox``, ``cls_to_id``,
``id_to_train``, ``part_id_to_part``, ``part_num``, ``id_to_part_loc``).
"""

import os


def draw_point(img, point, bbox_size=10, color=(0, 0, 255)):
    img[point[1] - bbox_size // 2: point[1] + bbox_size // 2,
        point[0] - bbox_size // 2: point[0] + bbox_size // 2] = color
    return img


def in_bbox(loc, bbox):
    """loc = (y, x); bbox = (y1, y2, x1, x2) inclusive."""
    return bbox[0] <= loc[0] <= bbox[1] and bbox[2] <= loc[1] <= bbox[3]


class CubPartAnnotations:
    def __init__(self, data_root: str):
        self.data_root = data_root
        self.id_to_path = {}
        self.id_to_bbox = {}
        self.cls_to_id = {}
        self.id_to_train = {}
        self.part_id_to_part = {}
        self.id_to_part_loc = {}

        with open(os.path.join(data_root, 'images.txt')) as f:
            for line in f:
                img_id, img_path = line.split(' ', 1)
                folder, name = img_path.strip().split('/')
                self.id_to_path[int(img_id)] = (folder, name)

        with open(os.path.join(data_root, 'bounding_boxes.txt')) as f:
            for line in f:
                cts = line.split(' ')
                img_id = int(cts[0])
                x, y, w, h = (int(float(c)) for c in cts[1:5])
                self.id_to_bbox[img_id] = (x, y, x + w, y + h)

        with open(os.path.join(data_root, 'image_class_labels.txt')) as f:
            for line in f:
                img_id, cls_id = line.split(' ')
                self.cls_to_id.setdefault(int(cls_id) - 1, []).append(int(img_id))

        split_path = os.path.join(data_root, 'train_test_split.txt')
        if os.path.isfile(split_path):
            with open(split_path) as f:
                for line in f:
                    img_id, is_train = line.split(' ')
                    self.id_to_train[int(img_id)] = int(is_train)

        with open(os.path.join(data_root, 'parts', 'parts.txt')) as f:
            for line in f:
                part_id, part_name = line.strip().split(' ', 1)
                self.part_id_to_part[part_id] = part_name
        self.part_num = len(self.part_id_to_part)

        with open(os.path.join(data_root, 'parts', 'part_locs.txt')) as f:
            for line in f:
                c = line.split(' ')
                img_id, part_id = int(c[0]), int(c[1])
                loc_x, loc_y, visible = int(float(c[2])), int(float(c[3])), int(c[4])
                self.id_to_part_loc.setdefault(img_id, [])
                if visible == 1:
                    self.id_to_part_loc[img_id].append([part_id, loc_x, loc_y])
