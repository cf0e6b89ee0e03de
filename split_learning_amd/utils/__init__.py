from .logger import Logger, print_with_color  # noqa: F401
