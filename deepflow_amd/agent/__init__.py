from .core import Agent  # noqa: F401
