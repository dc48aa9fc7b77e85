"""Dataset wrapper caching samples in the KV store
(reference: bagua/torch_api/contrib/cached_dataset.py:7-62)."""

from torch.utils.data.dataset import Dataset

from .cache_loader import CacheLoader

__all__ = ["CachedDataset"]


class CachedDataset(Dataset):
    """Caches ``dataset[i]`` in the backend store keyed
    ``{dataset_name}_{i}`` — worthwhile when __getitem__ is expensive
    (decode/augment) and the dataset fits the node's RAM."""

    def __init__(self, dataset: Dataset, backend: str = "tcp",
                 dataset_name: str = "", writer_buffer_size: int = 20,
                 **kwargs):
        self.dataset = dataset
        self.cache_loader = CacheLoader(backend, dataset_name,
                                        writer_buffer_size, **kwargs)

    def __getitem__(self, item):
        return self.cache_loader.get(item, lambda i: self.dataset[i])

    def __len__(self):
        return len(self.dataset)
