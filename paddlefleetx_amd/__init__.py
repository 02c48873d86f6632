"""MI355X-native large-model training & inference suite (PaddleFleetX-capability rebuild)."""

__version__ = "0.1.0"
