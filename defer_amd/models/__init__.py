from defer_amd.models.resnet import (DEFER_8STAGE_CUTS, resnet50,  # noqa: F401
                                     resnet101, resnet152)
from defer_amd.models.densenet import densenet121  # noqa: F401
from defer_amd.models.vgg import vgg19, vgg19_gap  # noqa: F401

MODELS = {"resnet50": resnet50, "resnet101": resnet101,
          "resnet152": resnet152, "vgg19": vgg19,
          "vgg19_gap": vgg19_gap, "densenet121": densenet121}
