from .loader import get_loader
from .transforms import build_train_and_test_transforms, TwoViewTransform

__all__ = ["get_loader", "build_train_and_test_transforms",
           "TwoViewTransform"]
