"""Model registry: name -> factory lookup (reference models/__init__.py:1-7).

The reference falls back to timm for archs missing from its registry
(trainer.py:123-128); this zoo implements every shipped baseline arch natively
(regnet/efficientnet included) and keeps the timm fallback for names outside
the zoo when timm happens to be installed.
"""

from .botnet import botnet50  # noqa: F401
from .densenet import densenet121, densenet161, densenet169, densenet201  # noqa: F401
from .efficientnet import efficientnet_b0  # noqa: F401
from .regnet import regnetx_160, regnety_160, regnety_320  # noqa: F401
from .resnet import (  # noqa: F401
    resnet18,
    resnet34,
    resnet50,
    resnet101,
    resnet152,
    resnext50_32x4d,
    resnext101_32x8d,
    wide_resnet50_2,
    wide_resnet101_2,
)


def build_model(arch, **kwargs):
    """Look the arch name up in this module's globals; unknown names fall
    back to ``timm.create_model`` when timm is installed (reference
    trainer.py:123-128 semantics), else raise KeyError."""
    try:
        factory = globals()[arch]
    except KeyError:
        try:
            import timm
        except ImportError:
            raise KeyError(
                f"Unknown arch '{arch}' (and timm is not installed). "
                "Available: "
                + ", ".join(sorted(k for k, v in globals().items()
                                   if callable(v)))
            ) from None
        nc = kwargs.pop("num_classes", 1000)
        return timm.create_model(arch, num_classes=nc, **kwargs)
    return factory(**kwargs)
