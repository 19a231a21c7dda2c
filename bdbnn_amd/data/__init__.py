from .loaders import (
    SyntheticImageDataset,
    LearnableSyntheticDataset,
    CIFAR10Dataset,
    CIFAR100Dataset,
    ImageFolderDataset,
    dataloader_cifar10,
    dataloader_cifar100,
    dataloader_imagenet,
    dataloader_synthetic,
)
