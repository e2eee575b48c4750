from .cnn import CNN2, CNN4, LeNet5, RefCNN6, build_model
from .resnet import BasicBlock, ResNet18

__all__ = ["CNN2", "CNN4", "RefCNN6", "LeNet5", "ResNet18", "BasicBlock",
           "build_model"]
