from .resnet import resnet50  # noqa: F401
from .res2net import res2net101  # noqa: F401
from .vgg import vgg16  # noqa: F401
