"""Built-in attacker registry.

String resolution matches the reference: attack name ``a`` maps to module
``blades_amd.attackers.<a>client`` and class ``<A>Client``
(reference: simulator.py:127-128).

``FUSABLE_CLIENT_TYPES`` lists the built-in clients whose training-time
behavior the fused many-model engine implements natively (label-flip target
map, sign-flip gradient negation); any user subclass falls back to the
per-client loop engine automatically.
"""
from __future__ import annotations

from .alieclient import AlieClient
from .ipmclient import IpmClient
from .labelflippingclient import LabelflippingClient
from .noiseclient import NoiseClient
from .signflippingclient import SignflippingClient

_REGISTRY = {
    "alie": AlieClient,
    "ipm": IpmClient,
    "labelflipping": LabelflippingClient,
    "noise": NoiseClient,
    "signflipping": SignflippingClient,
}

FUSABLE_CLIENT_TYPES = frozenset({
    AlieClient, IpmClient, NoiseClient,         # train honestly, attack post-gather
    LabelflippingClient, SignflippingClient,    # fused training-time semantics
})


def get_attacker_cls(name: str):
    try:
        return _REGISTRY[name.lower()]
    except KeyError:
        raise KeyError(f"unknown attack {name!r}; available: {sorted(_REGISTRY)}")


def register_attacker(name: str):
    def deco(cls):
        _REGISTRY[name.lower()] = cls
        return cls
    return deco


__all__ = [
    "AlieClient", "IpmClient", "LabelflippingClient", "NoiseClient",
    "SignflippingClient", "get_attacker_cls", "register_attacker",
    "FUSABLE_CLIENT_TYPES",
]
