"""Backbone feature extractors: ResNet-50 / Res2Net-101 / VGG-16 (5 stages)."""
from .resnet import resnet50  # noqa: F401
from .res2net import res2net101  # noqa: F401
from .vgg import vgg16  # noqa: F401
