from .sources import (AUGMENT_MODE_ENV, DataAugmenter, DataSource,
                      HFDatasetSource, ImageAugmenter, ImageFolderSource,
                      SyntheticImageSource, TensorSource, datasetMap,
                      register_dataset, register_image_folder)
from .records import (RecordShardDataset, RecordShardWriter, RecordSource,
                      register_record_dataset, write_records_from_source)
from .videos import VoxCeleb2LikeSource, lower_half_mask
from .dataloaders import (AugmentedDataset, CaptionDeletionTransform,
                          DevicePrefetcher, PrefetchLoader,
                          generate_collate_fn,
                          ShardedSampler,
                          collate_image_batch, get_dataset, get_dataset_online,
                          make_dataloader)

__all__ = [
    "AUGMENT_MODE_ENV", "DataAugmenter", "DataSource", "HFDatasetSource",
    "ImageAugmenter", "ImageFolderSource", "SyntheticImageSource",
    "TensorSource", "datasetMap", "register_dataset", "register_image_folder",
    "AugmentedDataset", "CaptionDeletionTransform", "DevicePrefetcher",
    "PrefetchLoader", "ShardedSampler", "generate_collate_fn",
    "collate_image_batch", "get_dataset", "get_dataset_online",
    "make_dataloader",
    "RecordShardDataset", "RecordShardWriter", "RecordSource",
    "register_record_dataset", "write_records_from_source",
    "VoxCeleb2LikeSource", "lower_half_mask",
]
