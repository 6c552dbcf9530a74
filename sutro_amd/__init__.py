"""sutro-amd: MI355X-native batch inference with the Sutro client API."""
__version__ = "0.1.0"
