from . import l7_schema  # noqa: F401
from .segment import L7Segment, SegmentSet  # noqa: F401
from .dictionary import TagDictionary  # noqa: F401
from .kg import KnowledgeGraphTable, KgInfo, default_platform  # noqa: F401
from .metrics import RollupTable, RollupFamily  # noqa: F401
