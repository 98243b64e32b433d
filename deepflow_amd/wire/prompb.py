"""Prometheus remote-write protobuf schemas (prompb WriteRequest subset;
reference vendors these under server/libs/datatype/prompb)."""

LABEL = {
    1: ("name", 's'),
    2: ("value", 's'),
}

SAMPLE = {
    1: ("value", 'd'),
    2: ("timestamp", 'i'),  # ms
}

TIMESERIES = {
    1: ("labels", '*m', LABEL),
    2: ("samples", '*m', SAMPLE),
}

WRITE_REQUEST = {
    1: ("timeseries", '*m', TIMESERIES),
}
