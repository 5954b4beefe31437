"""Benchmark model zoo for the reference's workloads:
VGG16 (the reference's headline end-to-end benchmark, README.md:52) and
ResNet-50 (BASELINE config 5).  Implemented natively — torchvision is not
part of the target image.
"""

from .resnet import ResNet50, resnet50
from .vgg import VGG16, vgg16

__all__ = ["VGG16", "vgg16", "ResNet50", "resnet50"]
