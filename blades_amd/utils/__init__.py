"""Utility layer: logging, seeding, metrics, model-weight helpers.

MI355X-native re-implementation of the reference utility surface
(reference: src/blades/utils.py:39-124) -- same public names
(``top1_accuracy``, ``accuracy``, ``initialize_logger``,
``reset_model_weights``, ``set_random_seed``) so user code written
against the reference keeps working, plus the framework-internal
helpers (tracing ranges, JSON stats logging).
"""
from .logging import initialize_logger, JsonStatsLogger
from .metrics import accuracy, top1_accuracy
from .seeding import set_random_seed, client_philox_seed
from .modeltools import reset_model_weights
from .tracing import trace_range, annotate

__all__ = [
    "initialize_logger",
    "JsonStatsLogger",
    "accuracy",
    "top1_accuracy",
    "set_random_seed",
    "client_philox_seed",
    "reset_model_weights",
    "trace_range",
    "annotate",
]
