from . import comfy_shim  # noqa: F401
