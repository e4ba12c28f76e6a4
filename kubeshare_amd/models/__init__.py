from .resnet import resnet18, resnet50
from .small import lstm, mnist_cnn
from .vgg import vgg16

MODEL_REGISTRY = {
    "resnet18": resnet18,
    "resnet50": resnet50,
    "vgg16": vgg16,
    # the reference's small e2e workload families (test/mnist,
    # test/tensorflow LSTM): 10-class heads
    "mnist": lambda n=10: mnist_cnn(10 if n == 1000 else n),
    "lstm": lambda n=10: lstm(10 if n == 1000 else n),
}


def build_model(name: str, num_classes: int = 1000):
    return MODEL_REGISTRY[name](num_classes)
