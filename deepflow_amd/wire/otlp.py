"""OTLP trace protobuf schemas (subset of opentelemetry-proto trace/v1 +
common/v1, wire-compatible field numbers; reference vendors the same protos
under message/opentelemetry)."""

ANY_VALUE = {
    1: ("string_value", 's'),
    2: ("bool_value", 'u'),
    3: ("int_value", 'i'),
    4: ("double_value", 'd'),
}

KEY_VALUE = {
    1: ("key", 's'),
    2: ("value", 'm', ANY_VALUE),
}

RESOURCE = {
    1: ("attributes", '*m', KEY_VALUE),
}

STATUS = {
    2: ("message", 's'),
    3: ("code", 'u'),  # 0 unset, 1 ok, 2 error
}

SPAN = {
    1: ("trace_id", 'b'),
    2: ("span_id", 'b'),
    3: ("trace_state", 's'),
    4: ("parent_span_id", 'b'),
    5: ("name", 's'),
    6: ("kind", 'u'),  # 0 unspec, 1 internal, 2 server, 3 client, 4 producer, 5 consumer
    7: ("start_time_unix_nano", 'x'),
    8: ("end_time_unix_nano", 'x'),
    9: ("attributes", '*m', KEY_VALUE),
    15: ("status", 'm', STATUS),
}

SCOPE = {
    1: ("name", 's'),
    2: ("version", 's'),
}

SCOPE_SPANS = {
    1: ("scope", 'm', SCOPE),
    2: ("spans", '*m', SPAN),
}

RESOURCE_SPANS = {
    1: ("resource", 'm', RESOURCE),
    2: ("scope_spans", '*m', SCOPE_SPANS),
}

TRACES_DATA = {
    1: ("resource_spans", '*m', RESOURCE_SPANS),
}

SPAN_KIND_SERVER = 2
SPAN_KIND_CLIENT = 3

# ---------------------------------------------------------------- logs/v1
LOG_RECORD = {
    1: ("time_unix_nano", 'x'),
    2: ("severity_number", 'u'),
    3: ("severity_text", 's'),
    5: ("body", 'm', ANY_VALUE),
    6: ("attributes", '*m', KEY_VALUE),
    9: ("trace_id", 'b'),
    10: ("span_id", 'b'),
    11: ("observed_time_unix_nano", 'x'),
}

SCOPE_LOGS = {
    1: ("scope", 'm', SCOPE),
    2: ("log_records", '*m', LOG_RECORD),
}

RESOURCE_LOGS = {
    1: ("resource", 'm', RESOURCE),
    2: ("scope_logs", '*m', SCOPE_LOGS),
}

LOGS_DATA = {
    1: ("resource_logs", '*m', RESOURCE_LOGS),
}

# ------------------------------------------------------------- metrics/v1
NUMBER_DATA_POINT = {
    2: ("start_time_unix_nano", 'x'),
    3: ("time_unix_nano", 'x'),
    4: ("as_double", 'd'),
    6: ("as_int", 'x'),  # sfixed64 on the wire
    7: ("attributes", '*m', KEY_VALUE),
}

GAUGE = {
    1: ("data_points", '*m', NUMBER_DATA_POINT),
}

SUM = {
    1: ("data_points", '*m', NUMBER_DATA_POINT),
    2: ("aggregation_temporality", 'u'),
    3: ("is_monotonic", 'u'),
}

HISTOGRAM_DATA_POINT = {
    2: ("start_time_unix_nano", 'x'),
    3: ("time_unix_nano", 'x'),
    4: ("count", 'x'),
    5: ("sum", 'd'),
    6: ("bucket_counts", '*x'),
    7: ("explicit_bounds", '*d'),
    9: ("attributes", '*m', KEY_VALUE),
}

HISTOGRAM = {
    1: ("data_points", '*m', HISTOGRAM_DATA_POINT),
    2: ("aggregation_temporality", 'u'),
}

METRIC = {
    1: ("name", 's'),
    2: ("description", 's'),
    3: ("unit", 's'),
    5: ("gauge", 'm', GAUGE),
    7: ("sum", 'm', SUM),
    9: ("histogram", 'm', HISTOGRAM),
}

SCOPE_METRICS = {
    1: ("scope", 'm', SCOPE),
    2: ("metrics", '*m', METRIC),
}

RESOURCE_METRICS = {
    1: ("resource", 'm', RESOURCE),
    2: ("scope_metrics", '*m', SCOPE_METRICS),
}

METRICS_DATA = {
    1: ("resource_metrics", '*m', RESOURCE_METRICS),
}
