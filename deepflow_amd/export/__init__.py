from .otlp_exporter import OtlpExporter  # noqa: F401
from .prom_exporter import PromExporter  # noqa: F401
