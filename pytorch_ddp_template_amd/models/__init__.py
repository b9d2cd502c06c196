from .foo import FooModel
from .resnet import ResNet, resnet18, resnet50
from .vit import ViT, vit_b16


def build_model(name: str, num_classes: int | None = None):
    """Factory used by the CLI / bench: name -> model instance."""
    name = name.lower()
    if name in ("foo", "foomodel", "mlp"):
        return FooModel()
    if name in ("resnet18", "resnet18-cifar"):
        return resnet18(num_classes or 10, stem="cifar")
    if name == "resnet18-imagenet":
        return resnet18(num_classes or 1000, stem="imagenet")
    if name == "resnet50":
        return resnet50(num_classes or 1000, stem="imagenet")
    if name in ("vit-b16", "vit_b16", "vitb16"):
        return vit_b16(num_classes or 1000)
    raise ValueError(f"unknown model {name!r}")


__all__ = [
    "FooModel",
    "ResNet",
    "resnet18",
    "resnet50",
    "ViT",
    "vit_b16",
    "build_model",
]
