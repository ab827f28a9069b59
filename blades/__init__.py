"""``blades`` — drop-in API alias for :mod:`blades_amd`.

Code written against the reference simulator imports ``blades.*``
(``from blades.simulator import Simulator``, ``import
blades.aggregators.median``, ``from blades.datasets import CIFAR10`` ...).
This package aliases every ``blades_amd`` submodule under the ``blades``
name — the SAME module objects, so class identities match across both
import paths and the reference's importlib string registries resolve.
"""
from __future__ import annotations

import importlib
import pkgutil
import sys

import blades_amd

_SKIP_PREFIXES = ("blades_amd._hip",)  # extensions load on demand


def _alias_all() -> None:
    sys.modules.setdefault("blades", sys.modules[__name__])
    for mod in pkgutil.walk_packages(blades_amd.__path__,
                                     prefix="blades_amd."):
        name = mod.name
        if name.startswith(_SKIP_PREFIXES):
            continue
        try:
            module = importlib.import_module(name)
        except ImportError:  # optional deps; keep the alias best-effort
            continue
        sys.modules["blades." + name[len("blades_amd."):]] = module


_alias_all()

from blades_amd import (BladesClient, ByzantineClient, BladesServer,  # noqa: E402,F401
                        Simulator, __version__)

__all__ = ["Simulator", "BladesClient", "ByzantineClient", "BladesServer",
           "__version__"]
