from . import datasets
from .datasets import (MNIST, FashionMNIST, CIFAR10, CIFAR100,
                       ImageRecordDataset, ImageFolderDataset,
                       SyntheticImageDataset)
from . import transforms
