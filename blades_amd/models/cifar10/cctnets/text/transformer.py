"""Reference-path module (reference: cctnets/text/transformer.py)."""
from ....text_cct import (text_transformer_2, text_transformer_4,  # noqa: F401
                          text_transformer_6)
