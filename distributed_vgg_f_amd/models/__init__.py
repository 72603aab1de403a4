"""Model zoo: VGG-F (the reference's model), VGG-16, ResNet-50."""

from __future__ import annotations

from ..config import Config
from .resnet50 import ResNet50
from .vgg16 import VGG16
from .vggf import VGGF

_REGISTRY = {
    "vggf": VGGF,
    "vgg16": VGG16,
    "resnet50": ResNet50,
}


def build_model(cfg: Config):
    try:
        cls = _REGISTRY[cfg.model]
    except KeyError:
        raise ValueError(f"unknown model {cfg.model!r}; have {sorted(_REGISTRY)}")
    return cls(cfg)
