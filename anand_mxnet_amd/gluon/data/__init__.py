from .dataset import (Dataset, ArrayDataset, SimpleDataset,
                      RecordFileDataset, SyntheticDataset)
from .sampler import Sampler, SequentialSampler, RandomSampler, BatchSampler
from .dataloader import DataLoader, default_batchify_fn
from . import vision
