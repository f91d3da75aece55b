"""Alias of the native Wide-ResNet (reference: keras_model_zoo/wresnet.py)."""
from theanompi_amd.models.wresnet import WResNet  # noqa: F401
