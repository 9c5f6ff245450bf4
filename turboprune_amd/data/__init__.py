from turboprune_amd.data.cifar import AirbenchLoaders, CifarLoader  # noqa: F401
from turboprune_amd.data.imagenet import (  # noqa: F401
    ImageNetLoaders,
    ShardedImageNet,
    SyntheticImageNet,
)
from turboprune_amd.data import augment  # noqa: F401
