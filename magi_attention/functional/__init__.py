from .flex_flash_attn import flex_flash_attn_func  # noqa: F401
