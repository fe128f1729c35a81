"""Importing this package registers every layer type."""

from . import vision, neuron, common, loss, data  # noqa: F401
from ..core.layer import LAYER_REGISTRY, create_layer, register_layer

__all__ = ["LAYER_REGISTRY", "create_layer", "register_layer"]
