from . import functional, primitive  # noqa: F401
from .primitive import WorkWithPostProcessFn, group_cast, group_reduce  # noqa: F401
