from .datasets import (CachedImageFolder, COCODetectionDataset,  # noqa: F401
                       ClassificationDataset, MosaicDetection,
                       SegmentationDataset, VOCDetectionDataset,
                       ZipImageDataset, mosaic4, read_split_data)
from .prefetcher import DataPrefetcher  # noqa: F401
from .samplers import (GroupedBatchSampler, InfiniteSampler,  # noqa: F401
                       SubsetRandomSampler, YoloBatchSampler,
                       create_aspect_ratio_groups)
from .synthetic import DeviceBatchLoader, SyntheticClassification  # noqa: F401
from .transforms import (ColorJitter, Compose, Mixup,  # noqa: F401
                         Normalize, RandomErasing, RandomResizedCrop,
                         classification_eval_transform,
                         classification_train_transform)
