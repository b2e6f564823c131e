from fl4health_amd.models.cnn import MnistNet, SmallCnn
from fl4health_amd.models.resnet import ResNet18

__all__ = ["SmallCnn", "MnistNet", "ResNet18"]
