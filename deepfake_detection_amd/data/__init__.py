from .config import resolve_data_config  # noqa: F401
from .constants import *  # noqa: F401,F403
from .auto_augment import (  # noqa: F401
    AutoAugment,
    RandAugment,
    AugMixAugment,
    auto_augment_transform,
    rand_augment_transform,
    augment_and_mix_transform,
)
from .dataset import (  # noqa: F401
    AugMixDataset,
    ConcatDataset,
    Dataset,
    DatasetTar,
    DeepFakeDataset_v1,
    DeepFakeDataset_v1_bak,
    DeepFakeDataset_v2,
    DeepFakeDataset_v3,
    SyntheticDeepFakeDataset,
)
from .distributed_sampler import OrderedDistributedSampler  # noqa: F401
from .loader import (  # noqa: F401
    PrefetchLoader,
    PrefetchLoader_v1,
    PrefetchLoader_v3,
    create_deepfake_loader,
    create_deepfake_loader_v1,
    create_deepfake_loader_v2,
    create_deepfake_loader_v3,
    create_loader,
    fast_collate,
    fast_collate_v1,
)
from .mixup import FastCollateMixup, mixup_batch, mixup_target, one_hot  # noqa: F401
from .random_erasing import RandomErasing  # noqa: F401
from .transforms_factory import (  # noqa: F401
    create_transform,
    transforms_deepfake_eval_v3,
    transforms_deepfake_train_v3,
)
