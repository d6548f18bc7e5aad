from torchbeast_amd.models.atari_net import AtariNet
from torchbeast_amd.models.resnet import ResNet

__all__ = ["AtariNet", "ResNet"]
