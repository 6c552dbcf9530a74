"""sutro-amd: MI355X-native batch LLM inference with the Sutro client API.

Module-level singleton + re-exports (reference `/root/reference/sutro/__init__.py`):

    import sutro_amd as so
    so.infer(["hello"], model="qwen-3-0.6b")
"""

from __future__ import annotations

__version__ = "0.1.0"

from .interfaces import JobStatus  # noqa: F401 (public API)
from .sdk import Sutro  # noqa: F401

_instance = Sutro()

# re-export every public method of the singleton as a module-level function
for _name in dir(_instance):
    if not _name.startswith("_") and callable(getattr(_instance, _name)):
        globals()[_name] = getattr(_instance, _name)
del _name
