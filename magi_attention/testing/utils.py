"""Env-var switching helpers for tests (reference surface:
testing/utils.py — switch_envvar_context / switch_envvar_decorator /
switch_envvars)."""
from __future__ import annotations

import os
from contextlib import contextmanager
from functools import wraps
from typing import Callable

from ..utils import wrap_to_list


@contextmanager
def switch_envvar_context(
    envvar_name: str | list[str],
    enable: bool = True,
    enable_value: str = "1",
    disable_value: str = "0",
):
    """Temporarily set the env var(s) to enable_value/disable_value,
    restoring (or deleting) on exit."""
    names = wrap_to_list(envvar_name)
    saved = [os.environ.get(n) for n in names]
    for n in names:
        os.environ[n] = enable_value if enable else disable_value
    try:
        yield
    finally:
        for n, old in zip(names, saved):
            if old is None:
                os.environ.pop(n, None)
            else:
                os.environ[n] = old


def switch_envvar_decorator(
    envvar_name: str | list[str] | None = None,
    enable: bool = True,
    enable_value: str = "1",
    disable_value: str = "0",
) -> Callable:
    def decorator(func: Callable) -> Callable:
        if envvar_name is None:
            return func

        @wraps(func)
        def wrapper(*args, **kwargs):
            with switch_envvar_context(
                envvar_name, enable, enable_value, disable_value
            ):
                return func(*args, **kwargs)

        return wrapper

    return decorator


def switch_envvars(
    envvar_name_list: list[str],
    enable_dict: dict[str, bool] = {},
    enable_value_dict: dict[str, str] = {},
    disable_value_dict: dict[str, str] = {},
) -> Callable[[], None]:
    """Set several env vars at once; returns a callback restoring them all
    (in reverse order)."""
    saved = [(n, os.environ.get(n)) for n in envvar_name_list]
    for n in envvar_name_list:
        if enable_dict.get(n, True):
            os.environ[n] = enable_value_dict.get(n, "1")
        else:
            os.environ[n] = disable_value_dict.get(n, "0")

    def switch_envvars_back() -> None:
        for n, old in reversed(saved):
            if old is None:
                os.environ.pop(n, None)
            else:
                os.environ[n] = old

    return switch_envvars_back
