"""Reference-path module (reference: cctnets/text/cct.py)."""
from ....text_cct import (TextCCT, text_cct_2, text_cct_4,  # noqa: F401
                          text_cct_6)
