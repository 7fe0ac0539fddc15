from .toy import HipLinear, toy_model  # noqa: F401
from .resnet import resnet50, ResNet  # noqa: F401
