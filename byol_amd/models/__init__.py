from .resnet import ResNetEncoder, build_encoder, arch_names
from .byol import BYOL, CosEMA, FlatParamSpace

__all__ = ["ResNetEncoder", "build_encoder", "arch_names", "BYOL", "CosEMA",
           "FlatParamSpace"]
