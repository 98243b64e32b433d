from .otlp_exporter import OtlpExporter  # noqa: F401
