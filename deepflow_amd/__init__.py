"""deepflow_amd — MI355X-native observability backplane.

A ground-up rebuild of the capabilities of deepflowio/deepflow for AMD
Instinct MI355X nodes: the agent-facing protobuf wire protocol and framing are
byte-compatible with the reference, while the server-side hot path (ingest
decode, SmartEncoding tag-dictionary join, columnar hot store, query
group-by/aggregation) runs as hand-written CDNA4 HIP kernels (gfx950) over
HBM-resident columnar segments, sharded across up to 8 GPUs with RCCL
collectives over xGMI.

Layers (see SURVEY.md for the reference blueprint):
  wire/      protobuf schemas + trident framing (ABI with reference agents)
  gen/       synthetic flow/span/metric generators (fixtures + bench drivers)
  ops/       HIP kernels (csrc/*.hip) + ctypes bindings + CPU references
  ingest/    receiver -> GPU decode -> SmartEncoding -> store pipelines
  store/     GPU columnar store, tag dictionaries, flow_tag tables
  query/     DF-SQL engine, PromQL, Tempo, profile APIs over the hot store
  parallel/  multi-GPU sharding + RCCL dictionary sync / bucket reduce
  agent/     host-side collection (flow map, L7 parsing, sender)
  control/   controller-lite (agent registry, platform data, tagrecorder)
  utils/     self-telemetry counters, config
"""

__version__ = "0.1.0"
