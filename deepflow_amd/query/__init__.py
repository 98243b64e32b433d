from .engine import QueryEngine  # noqa: F401
from .sql import parse_sql, SqlError  # noqa: F401
