"""Layer catalog.  Importing this package registers every layer type."""

from . import data, recurrent, structure, vision  # noqa: F401
from .base import LAYER_REGISTRY, Layer, create_layer, register_layer  # noqa: F401
