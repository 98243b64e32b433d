from .spans import SpanGenConfig, gen_span_dict, gen_span_payload  # noqa: F401
from .flows import FlowGenConfig, gen_flow_dict, gen_flow_payload  # noqa: F401
from .documents import DocGenConfig, gen_document_dict, gen_document_payload  # noqa: F401
