from .interface import TFieldClient  # noqa: F401
