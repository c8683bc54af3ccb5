from . import native, torch_ref  # noqa: F401
