from .functional import (add_relu, batchnorm2d, conv2d, conv_relu_pool,
                         dense_head2, global_avgpool, linear, maxpool,
                         maxpool2x2, softmax_xent)
from .modules import (BatchNorm2d, Conv2dValid, Dense, Flatten,
                      GlobalAvgPool, MaxPool, MaxPool2x2)
from .adam import FusedAdam

__all__ = ["conv2d", "conv_relu_pool", "dense_head2", "maxpool2x2",
           "maxpool", "linear", "softmax_xent",
           "batchnorm2d", "global_avgpool", "add_relu",
           "Conv2dValid", "MaxPool2x2", "MaxPool", "Dense", "Flatten",
           "BatchNorm2d", "GlobalAvgPool", "FusedAdam"]
