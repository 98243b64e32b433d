from .stats import Counter, StatsRegistry, default_registry  # noqa: F401
