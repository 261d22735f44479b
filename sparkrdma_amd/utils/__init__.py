from .tensors import as_device_i64  # noqa: F401
