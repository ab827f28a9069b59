"""Reference-path package: ``blades.models.cifar10.cctnets`` — the
vendored Compact-Transformers library in the reference; here a facade
over the single-file rewrites (models/cct.py, models/text_cct.py)."""
from ...cct import CCT, cct_2_3x2_32, cvt_2_4_32, vit_lite_2_4_32  # noqa: F401
from . import cct, cvt, vit  # noqa: F401
