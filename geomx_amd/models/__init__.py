from .cnn import GeoCNN, geo_cnn  # noqa: F401
from .resnet import ResNet, resnet50  # noqa: F401


def create_model(name: str, **kwargs):
    name = name.lower()
    if name in ("geomx_cnn", "geomx-cnn", "cnn"):
        return geo_cnn(**kwargs)
    if name in ("resnet50", "resnet-50"):
        kwargs.pop("image_size", None)  # ResNet is size-agnostic
        return resnet50(**kwargs)
    raise ValueError(f"unknown model {name!r}")
