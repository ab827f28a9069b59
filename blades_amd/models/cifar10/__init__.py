"""Reference-path package: ``blades.models.cifar10``
(reference: src/blades/models/cifar10/__init__.py exposing CCTNet)."""
from ..cct import CCT, CCTNet, cct_2_3x2_32, create_model  # noqa: F401
from ..resnet import resnet18, wide_resnet28_10  # noqa: F401
