"""Config + API-key discovery (reference: `/root/reference/sutro/validation.py`).

Config lives at ~/.sutro/config.json ({"api_key": ..., "base_url": ...}).
There is no network in this environment, so the reference's PyPI version nag
is replaced by a no-op local version check.
"""

from __future__ import annotations

import json
import os
from typing import Optional

CONFIG_DIR = os.path.expanduser("~/.sutro")
CONFIG_PATH = os.path.join(CONFIG_DIR, "config.json")


def load_config() -> dict:
    try:
        with open(CONFIG_PATH) as f:
            return json.load(f)
    except (FileNotFoundError, json.JSONDecodeError):
        return {}


def save_config(config: dict) -> None:
    os.makedirs(CONFIG_DIR, exist_ok=True)
    with open(CONFIG_PATH, "w") as f:
        json.dump(config, f, indent=2)


def check_for_api_key() -> Optional[str]:
    """ENV first, then config file (reference `validation.py:36-60`)."""
    key = os.environ.get("SUTRO_API_KEY")
    if key:
        return key
    return load_config().get("api_key")


def check_version() -> None:
    """Offline no-op (the reference pings PyPI, `validation.py:10-33`)."""
    return None
