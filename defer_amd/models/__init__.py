from defer_amd.models.resnet import resnet50, DEFER_8STAGE_CUTS  # noqa: F401
from defer_amd.models.vgg import vgg19  # noqa: F401
