"""Alias of the native VGG-16 (reference: lasagne_model_zoo/vgg.py)."""
from theanompi_amd.models.vgg16 import VGG16  # noqa: F401
