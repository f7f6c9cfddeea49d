from .cil_model import CilModel, CilClassifier, get_backbone, freeze_parameters
from .resnet_cifar import (CifarResNet, resnet20, resnet32, resnet44, resnet56,
                           resnet110)
from .resnet import ResNet, resnet18, resnet34, resnet50

__all__ = ["CilModel", "CilClassifier", "get_backbone", "freeze_parameters",
           "CifarResNet", "ResNet", "resnet20", "resnet32", "resnet44", "resnet56",
           "resnet110", "resnet18", "resnet34", "resnet50"]
