from .cnn import CNN2, CNN4, RefCNN6, LeNet5, build_model

__all__ = ["CNN2", "CNN4", "RefCNN6", "LeNet5", "build_model"]
