"""Reference-path package (reference: cctnets/text/)."""
from . import cct, transformer  # noqa: F401
