from . import pb, flow_log, metric, framing, const_enums  # noqa: F401
