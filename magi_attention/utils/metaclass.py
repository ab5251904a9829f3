"""Generic metaclass helpers (reference surface: utils/metaclass.py)."""
from __future__ import annotations

import threading
from typing import Any


class SingletonMeta(type):
    """Thread-safe singleton metaclass: every construction of a class using
    it returns the one shared instance (double-checked under an RLock)."""

    _instances: dict[type, Any] = {}
    _lock = threading.RLock()

    def __call__(cls, *args, **kwargs):
        if cls not in SingletonMeta._instances:
            with SingletonMeta._lock:
                if cls not in SingletonMeta._instances:
                    SingletonMeta._instances[cls] = super().__call__(*args, **kwargs)
        return SingletonMeta._instances[cls]
