from .l7_pipeline import L7IngestPipeline  # noqa: F401
