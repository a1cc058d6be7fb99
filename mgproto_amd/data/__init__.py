from .synthetic import SyntheticImages, DeviceBatchPool  # noqa: F401
