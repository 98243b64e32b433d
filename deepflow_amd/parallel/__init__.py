from .dict_sync import DictSync  # noqa: F401
