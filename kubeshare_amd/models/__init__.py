from .resnet import resnet18, resnet50
from .vgg import vgg16

MODEL_REGISTRY = {
    "resnet18": resnet18,
    "resnet50": resnet50,
    "vgg16": vgg16,
}


def build_model(name: str, num_classes: int = 1000):
    return MODEL_REGISTRY[name](num_classes)
