"""Reference-path module (reference: cctnets/cct.py)."""
from ...cct import CCT, cct_2_3x2_32, create_model  # noqa: F401
