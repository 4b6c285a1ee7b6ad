from .resnet import ResNet, BasicBlock, Bottleneck, resnet18, resnet34, resnet50, build_model  # noqa: F401
from .mobilenet import mobilenet_v2  # noqa: F401
from .partition import partition_model, partition_units, split_counts  # noqa: F401
from .layers import (ConvBNAct, DepthwiseConvBNAct, Linear, GlobalAvgPool,  # noqa: F401
                     MaxPool2d3x3s2, refresh_all_shadows)
