from sparktorch_amd.models.mnist import MnistCNN, MnistMLP
from sparktorch_amd.models.resnet import ResNet18
from sparktorch_amd.models.simple_net import (
    AutoEncoder,
    ClassificationNet,
    Net,
    NetworkWithParameters,
)
