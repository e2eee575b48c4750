from .functional import conv2d, maxpool2x2, linear, softmax_xent
from .modules import Conv2dValid, MaxPool2x2, Dense, Flatten
from .adam import FusedAdam

__all__ = ["conv2d", "maxpool2x2", "linear", "softmax_xent",
           "Conv2dValid", "MaxPool2x2", "Dense", "Flatten", "FusedAdam"]
