from turboprune_amd.models.wrappers import (  # noqa: F401
    PruneModel,
    available_models,
    build_model,
    num_classes_of,
)
from turboprune_amd.models.resnet import (  # noqa: F401
    resnet18, resnet34, resnet50, resnet101,
)
from turboprune_amd.models.vgg import vgg11_bn, vgg16_bn, vgg19_bn  # noqa: F401
from turboprune_amd.models import deit  # noqa: F401
