from msbn.models.simple import SimpleCNN  # noqa: F401
from msbn.models.resnet import resnet18, resnet50, ResNet  # noqa: F401
from msbn.models.dcgan import Generator, Discriminator  # noqa: F401

__all__ = [
    "SimpleCNN",
    "resnet18",
    "resnet50",
    "ResNet",
    "Generator",
    "Discriminator",
]


def retinanet(num_classes: int = 80):
    from msbn.models.retinanet import RetinaNet

    return RetinaNet(num_classes)
