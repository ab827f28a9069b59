"""Reference-path module (reference: cctnets/cvt.py)."""
from ...cct import CCT, cvt_2_4_32  # noqa: F401
