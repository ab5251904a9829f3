"""Meta-result collections (reference surface: meta/collection/ —
DispatchMeta, CalcMeta, CommMeta). This rebuild defines the classes in
meta/__init__.py (DispatchMeta) and meta/containers.py (CalcMeta, CommMeta);
this module re-exports them under the reference's package path."""
from ..containers import CalcMeta, CommMeta


def __getattr__(name):
    # DispatchMeta lives in the meta package root; imported lazily to avoid
    # a circular import during meta/__init__ execution
    if name == "DispatchMeta":
        from .. import DispatchMeta

        return DispatchMeta
    raise AttributeError(name)


__all__ = ["DispatchMeta", "CalcMeta", "CommMeta"]
