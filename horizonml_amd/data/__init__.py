from .cifar import (SyntheticCIFAR10, CIFAR10Local, build_dataset,  # noqa: F401
                    get_dataloader)
