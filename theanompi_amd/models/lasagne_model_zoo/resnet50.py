"""Alias of the native ResNet-50 (reference: lasagne_model_zoo/resnet50.py)."""
from theanompi_amd.models.resnet50 import ResNet50  # noqa: F401
