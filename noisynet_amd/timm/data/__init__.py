from .auto_augment import (AutoAugment, RandAugment,  # noqa: F401
                           auto_augment_transform, rand_augment_transform)
from .config import resolve_data_config  # noqa: F401
from .dataset import Dataset, DatasetTar, SyntheticImageDataset  # noqa: F401
from .distributed_sampler import OrderedDistributedSampler  # noqa: F401
from .loader import PrefetchLoader, create_loader, fast_collate  # noqa: F401
from .mixup import FastCollateMixup, mixup_batch, mixup_target  # noqa: F401
from .random_erasing import RandomErasing  # noqa: F401
from .transforms import (transforms_imagenet_eval,  # noqa: F401
                         transforms_imagenet_train)
