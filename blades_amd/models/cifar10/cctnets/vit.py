"""Reference-path module (reference: cctnets/vit.py)."""
from ...cct import CCT, vit_lite_2_4_32  # noqa: F401
