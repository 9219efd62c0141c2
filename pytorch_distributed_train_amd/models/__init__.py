from .resnet import (resnet18, resnet34, resnet50, resnet101,
                     resnet152)
from . import layers


def build_model(name: str, num_classes: int, compute_dtype):
    if name == "resnet18":
        return resnet18(num_classes=num_classes, compute_dtype=compute_dtype)
    if name == "resnet34":
        return resnet34(num_classes=num_classes, compute_dtype=compute_dtype)
    if name == "resnet50":
        return resnet50(num_classes=num_classes, compute_dtype=compute_dtype)
    if name == "resnet101":
        return resnet101(num_classes=num_classes,
                         compute_dtype=compute_dtype)
    if name == "resnet152":
        return resnet152(num_classes=num_classes, compute_dtype=compute_dtype)
    if name == "vit_b_16":
        from .vit import vit_b_16
        return vit_b_16(num_classes=num_classes, compute_dtype=compute_dtype)
    raise ValueError(f"unknown model {name}")


__all__ = ["build_model", "resnet18", "resnet34", "resnet50",
           "resnet101", "resnet152", "layers"]
