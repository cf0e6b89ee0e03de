from .loader import data_loader  # noqa: F401
