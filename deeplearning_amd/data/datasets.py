"""Datasets: folder-split classification, VOC XML and COCO JSON detection,
segmentation mask pairs — PIL-based, no torchvision.

Reference parity: classification/mnist/dataLoader/{dataSet,dataLoader}.py
(read_split_data, MyDataSet), detection/fasterRcnn/dataLoader/voc_dataset.py,
detection/YOLOX/yolox/data/datasets/{voc,coco}.py,
mosaic: detection/yolov5/utils/datasets.py:776-847.
"""
from __future__ import annotations

import json
import os
import random
import xml.etree.ElementTree as ET
from pathlib import Path

import torch
from PIL import Image
from torch.utils.data import Dataset

from .transforms import pil_to_tensor

IMG_EXTS = {".jpg", ".jpeg", ".png", ".bmp", ".webp"}


def read_split_data(root: str, val_rate: float = 0.2, seed: int = 0):
    """Walk class-per-folder root -> (train_paths, train_labels, val_paths,
    val_labels, class_names) (ref classification/mnist/dataLoader/dataSet.py)."""
    root = Path(root)
    classes = sorted(d.name for d in root.iterdir() if d.is_dir())
    class_idx = {c: i for i, c in enumerate(classes)}
    rng = random.Random(seed)
    train_p, train_l, val_p, val_l = [], [], [], []
    for c in classes:
        imgs = sorted(p for p in (root / c).iterdir()
                      if p.suffix.lower() in IMG_EXTS)
        val = set(rng.sample(range(len(imgs)), int(len(imgs) * val_rate)))
        for i, p in enumerate(imgs):
            if i in val:
                val_p.append(str(p))
                val_l.append(class_idx[c])
            else:
                train_p.append(str(p))
                train_l.append(class_idx[c])
    return train_p, train_l, val_p, val_l, classes


class ClassificationDataset(Dataset):
    def __init__(self, paths, labels, transform=None):
        self.paths = paths
        self.labels = labels
        self.transform = transform

    def __len__(self):
        return len(self.paths)

    def __getitem__(self, i):
        img = Image.open(self.paths[i]).convert("RGB")
        if self.transform:
            img = self.transform(img)
        else:
            img = pil_to_tensor(img)
        return img, self.labels[i]

    @staticmethod
    def collate_fn(batch):
        imgs, labels = zip(*batch)
        return torch.stack(imgs), torch.tensor(labels)


class VOCDetectionDataset(Dataset):
    """Pascal-VOC layout: JPEGImages/ + Annotations/*.xml + ImageSets/Main
    (ref detection/fasterRcnn/dataLoader/voc_dataset.py)."""

    def __init__(self, root, image_set="train", transforms=None,
                 class_names=None):
        self.root = Path(root)
        split = self.root / "ImageSets" / "Main" / f"{image_set}.txt"
        self.ids = [l.strip() for l in open(split) if l.strip()]
        self.transforms = transforms
        self.class_names = class_names or VOC_CLASSES
        self.class_idx = {c: i + 1 for i, c in enumerate(self.class_names)}

    def __len__(self):
        return len(self.ids)

    def parse_xml(self, xml_path):
        tree = ET.parse(xml_path)
        boxes, labels, iscrowd = [], [], []
        for obj in tree.findall("object"):
            name = obj.find("name").text
            if name not in self.class_idx:
                continue
            bb = obj.find("bndbox")
            box = [float(bb.find(k).text)
                   for k in ("xmin", "ymin", "xmax", "ymax")]
            if box[2] <= box[0] or box[3] <= box[1]:
                continue
            boxes.append(box)
            labels.append(self.class_idx[name])
            diff = obj.find("difficult")
            iscrowd.append(int(diff.text) if diff is not None else 0)
        return (torch.tensor(boxes, dtype=torch.float32).reshape(-1, 4),
                torch.tensor(labels, dtype=torch.int64),
                torch.tensor(iscrowd, dtype=torch.int64))

    def get_height_and_width(self, i):
        """Read (h, w) from the XML only — lets GroupedBatchSampler group by
        aspect ratio without decoding images (ref fasterRcnn voc_dataset)."""
        tree = ET.parse(self.root / "Annotations" / f"{self.ids[i]}.xml")
        size = tree.find("size")
        if size is not None:
            return (int(size.find("height").text),
                    int(size.find("width").text))
        img = Image.open(self.root / "JPEGImages" / f"{self.ids[i]}.jpg")
        return img.height, img.width

    def __getitem__(self, i):
        img_id = self.ids[i]
        img = Image.open(self.root / "JPEGImages" / f"{img_id}.jpg")
        img = pil_to_tensor(img.convert("RGB"))
        boxes, labels, iscrowd = self.parse_xml(
            self.root / "Annotations" / f"{img_id}.xml")
        target = {"boxes": boxes, "labels": labels, "iscrowd": iscrowd,
                  "image_id": torch.tensor([i])}
        if self.transforms:
            img, target = self.transforms(img, target)
        return img, target

    @staticmethod
    def collate_fn(batch):
        return tuple(zip(*batch))


VOC_CLASSES = ("aeroplane", "bicycle", "bird", "boat", "bottle", "bus", "car",
               "cat", "chair", "cow", "diningtable", "dog", "horse",
               "motorbike", "person", "pottedplant", "sheep", "sofa", "train",
               "tvmonitor")


class COCODetectionDataset(Dataset):
    """COCO instances JSON (ref YOLOX yolox/data/datasets/coco.py), parsed
    without pycocotools."""

    def __init__(self, img_dir, ann_file, transforms=None):
        self.img_dir = Path(img_dir)
        with open(ann_file) as f:
            coco = json.load(f)
        self.images = {im["id"]: im for im in coco["images"]}
        cats = sorted(c["id"] for c in coco["categories"])
        self.cat_idx = {cid: i + 1 for i, cid in enumerate(cats)}
        self.anns = {}
        for a in coco.get("annotations", []):
            self.anns.setdefault(a["image_id"], []).append(a)
        self.ids = sorted(self.images)
        self.transforms = transforms

    def __len__(self):
        return len(self.ids)

    def __getitem__(self, i):
        img_id = self.ids[i]
        info = self.images[img_id]
        img = Image.open(self.img_dir / info["file_name"]).convert("RGB")
        img = pil_to_tensor(img)
        boxes, labels, iscrowd = [], [], []
        for a in self.anns.get(img_id, []):
            x, y, w, h = a["bbox"]
            if w <= 0 or h <= 0:
                continue
            boxes.append([x, y, x + w, y + h])
            labels.append(self.cat_idx[a["category_id"]])
            iscrowd.append(a.get("iscrowd", 0))
        target = {"boxes": torch.tensor(boxes,
                                        dtype=torch.float32).reshape(-1, 4),
                  "labels": torch.tensor(labels, dtype=torch.int64),
                  "iscrowd": torch.tensor(iscrowd, dtype=torch.int64),
                  "image_id": torch.tensor([img_id])}
        if self.transforms:
            img, target = self.transforms(img, target)
        return img, target

    collate_fn = staticmethod(VOCDetectionDataset.collate_fn)


class SegmentationDataset(Dataset):
    """(image, mask) file pairs; masks are palette/grayscale PNGs
    (ref Image_segmentation/U-Net/dataLoader, FCN)."""

    def __init__(self, image_paths, mask_paths, transforms=None):
        assert len(image_paths) == len(mask_paths)
        self.images = image_paths
        self.masks = mask_paths
        self.transforms = transforms

    def __len__(self):
        return len(self.images)

    def __getitem__(self, i):
        img = pil_to_tensor(Image.open(self.images[i]).convert("RGB"))
        mask = torch.from_numpy(
            __import__("numpy").array(Image.open(self.masks[i]))).long()
        if self.transforms:
            img, mask = self.transforms(img, mask)
        return img, mask


def mosaic4(images, targets, out_size=640):
    """4-image mosaic: place 4 images around a random center
    (ref detection/yolov5/utils/datasets.py:776-847; tensor-level redesign).

    images: list of 4 CHW tensors; targets: list of dicts with xyxy 'boxes'.
    """
    s = out_size
    yc = random.randint(s // 4, 3 * s // 4)
    xc = random.randint(s // 4, 3 * s // 4)
    canvas = images[0].new_full((3, s, s), 0.447)
    all_boxes, all_labels = [], []
    for i, (img, t) in enumerate(zip(images, targets)):
        c, h, w = img.shape
        if i == 0:   # top-left
            x1, y1, x2, y2 = max(xc - w, 0), max(yc - h, 0), xc, yc
            sx1, sy1 = w - (x2 - x1), h - (y2 - y1)
        elif i == 1:  # top-right
            x1, y1, x2, y2 = xc, max(yc - h, 0), min(xc + w, s), yc
            sx1, sy1 = 0, h - (y2 - y1)
        elif i == 2:  # bottom-left
            x1, y1, x2, y2 = max(xc - w, 0), yc, xc, min(yc + h, s)
            sx1, sy1 = w - (x2 - x1), 0
        else:        # bottom-right
            x1, y1, x2, y2 = xc, yc, min(xc + w, s), min(yc + h, s)
            sx1, sy1 = 0, 0
        canvas[:, y1:y2, x1:x2] = img[:, sy1:sy1 + (y2 - y1),
                                      sx1:sx1 + (x2 - x1)]
        if t["boxes"].numel():
            b = t["boxes"].clone()
            b[:, [0, 2]] += x1 - sx1
            b[:, [1, 3]] += y1 - sy1
            b[:, [0, 2]] = b[:, [0, 2]].clamp(0, s)
            b[:, [1, 3]] = b[:, [1, 3]].clamp(0, s)
            keep = (b[:, 2] - b[:, 0] > 2) & (b[:, 3] - b[:, 1] > 2)
            all_boxes.append(b[keep])
            all_labels.append(t["labels"][keep])
    boxes = torch.cat(all_boxes) if all_boxes else images[0].new_zeros((0, 4))
    labels = torch.cat(all_labels) if all_labels else \
        torch.zeros(0, dtype=torch.int64)
    return canvas, {"boxes": boxes, "labels": labels}


class MosaicDetection(Dataset):
    """Wrap a detection dataset with 4-image mosaic augmentation
    (ref detection/YOLOX/yolox/data/datasets/mosaicdetection.py). Each item
    draws 3 extra random indices and composes mosaic4; `enabled=False` (or a
    YoloBatchSampler (mosaic, idx) tuple with mosaic=False) passes through —
    the trainer flips it off for the last no-aug epochs."""

    def __init__(self, dataset, out_size=640, enabled=True):
        self.dataset = dataset
        self.out_size = out_size
        self.enabled = enabled

    def __len__(self):
        return len(self.dataset)

    def __getitem__(self, index):
        enabled = self.enabled
        if isinstance(index, tuple):  # YoloBatchSampler item
            enabled, index = index
        if not enabled:
            return self.dataset[index]
        idxs = [index] + [random.randrange(len(self.dataset))
                          for _ in range(3)]
        items = [self.dataset[i] for i in idxs]
        return mosaic4([im for im, _ in items], [t for _, t in items],
                       self.out_size)

    collate_fn = staticmethod(VOCDetectionDataset.collate_fn)


class CachedImageFolder(Dataset):
    """Class-per-folder dataset with in-memory caching.

    Reference parity: classification/swin_transformer/dataLoader/
    cached_image_folder.py:64 — cache_mode: 'no' (read from disk every time),
    'full' (decode once, keep tensors), 'part' (each worker caches its shard
    lazily).
    """

    def __init__(self, root, transform=None, cache_mode="no", val_rate=0.0,
                 split="train", seed=0):
        tp, tl, vp, vl, self.classes = read_split_data(root, val_rate, seed)
        self.paths, self.labels = (tp, tl) if split == "train" else (vp, vl)
        self.transform = transform
        assert cache_mode in ("no", "part", "full")
        self.cache_mode = cache_mode
        self._cache = {}
        if cache_mode == "full":
            for i in range(len(self.paths)):
                self._cache[i] = self._load(i)

    def _load(self, i):
        return Image.open(self.paths[i]).convert("RGB")

    def __len__(self):
        return len(self.paths)

    def __getitem__(self, i):
        if self.cache_mode == "no":
            img = self._load(i)
        elif i in self._cache:
            img = self._cache[i]
        else:
            img = self._load(i)
            self._cache[i] = img  # 'part': lazily fill this worker's shard
        out = self.transform(img) if self.transform else pil_to_tensor(img)
        return out, self.labels[i]


class ZipImageDataset(Dataset):
    """Images inside a zip archive + a (member, label) index list.

    Reference parity: swin dataLoader/zipreader.py:23 (ZipReader) — one zip
    handle per worker process (zipfile handles are not fork-safe).
    """

    def __init__(self, zip_path, index, transform=None):
        """index: list of (member_name, label) pairs."""
        self.zip_path = str(zip_path)
        self.index = list(index)
        self.transform = transform
        self._zf = None
        self._pid = None

    def _zip(self):
        pid = os.getpid()
        if self._zf is None or self._pid != pid:
            import zipfile

            self._zf = zipfile.ZipFile(self.zip_path, "r")
            self._pid = pid
        return self._zf

    def __len__(self):
        return len(self.index)

    def __getitem__(self, i):
        name, label = self.index[i]
        import io

        with self._zip().open(name) as f:
            img = Image.open(io.BytesIO(f.read())).convert("RGB")
        out = self.transform(img) if self.transform else pil_to_tensor(img)
        return out, label
