from .cifar import (PickleCIFAR100, SyntheticCIFAR, build_datasets,
                    build_loaders, normalize, random_crop_padded)

__all__ = ["PickleCIFAR100", "SyntheticCIFAR", "build_datasets",
           "build_loaders", "normalize", "random_crop_padded"]
