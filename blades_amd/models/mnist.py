"""Reference-path module: ``blades.models.mnist``
(reference: src/blades/models/mnist/__init__.py exposing MLP/DNN)."""
from .mlp import MLP, create_model  # noqa: F401

DNN = MLP  # the reference file was named dnn.py
