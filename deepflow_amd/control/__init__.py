from .controller import ControllerLite  # noqa: F401
