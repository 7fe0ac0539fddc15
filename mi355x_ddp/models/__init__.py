from .toy import HipLinear, toy_model  # noqa: F401
from .resnet import resnet50, ResNet  # noqa: F401
from .vit import vit_l_32, vit_tiny, VisionTransformer  # noqa: F401
