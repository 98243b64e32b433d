"""Agent configuration: template.yaml-style defaults + controller diffs.

The reference agent ships a config template (agent/config/template.yaml)
whose values arrive both from the local file and from the controller's
versioned Config push; modules register diff callbacks and react only to
keys that changed. Same model here: `AgentConfig.apply` merges a new
config dict, computes the changed keys, and fires the registered
callbacks for their prefixes.
"""
from __future__ import annotations

import os
from typing import Any, Callable, Dict, List, Optional

import yaml

# template defaults (subset of the reference's template.yaml surface our
# agent acts on)
TEMPLATE: Dict[str, Any] = {
    "max_memory": 768,           # MiB, guard threshold
    "sync_interval": 60,         # s
    "stats_interval": 10,
    "global_pps_threshold": 200000,
    "tap_interface_regex": "^(tap.*|cali.*|veth.*|eth.*|en[ospx].*|lo)",
    "capture_packet_size": 65535,
    "l7_log_packet_size": 1024,
    "l4_log_tap_types": [0],
    "l7_protocol_enabled": [],   # empty = all
    "custom_protocol_ports": [],
    "flow_timeout": 60,
    "compressor_socket_type": "zstd",
    "max_flows": 1 << 20,
}


class AgentConfig:
    def __init__(self, data: Optional[Dict[str, Any]] = None):
        self.values: Dict[str, Any] = dict(TEMPLATE)
        if data:
            self.values.update(data)
        self.version = 0
        self._callbacks: List = []  # (key_prefix, fn(key, old, new))

    @classmethod
    def load(cls, path: Optional[str] = None) -> "AgentConfig":
        path = path or os.environ.get("DEEPFLOW_AGENT_CONFIG",
                                      "/etc/deepflow-agent.yaml")
        data: Dict[str, Any] = {}
        if os.path.exists(path):
            with open(path) as f:
                data = yaml.safe_load(f) or {}
        return cls(data)

    def on_change(self, key_prefix: str,
                  fn: Callable[[str, Any, Any], None]) -> None:
        self._callbacks.append((key_prefix, fn))

    def apply(self, new_values: Dict[str, Any],
              version: Optional[int] = None) -> List[str]:
        """Merge a pushed config; fire callbacks for changed keys only.
        Returns the changed key list."""
        changed = []
        for k, v in new_values.items():
            old = self.values.get(k)
            if old != v:
                self.values[k] = v
                changed.append(k)
                for prefix, fn in self._callbacks:
                    if k.startswith(prefix):
                        fn(k, old, v)
        if version is not None:
            self.version = version
        elif changed:
            self.version += 1
        return changed

    def get(self, key: str, default: Any = None) -> Any:
        return self.values.get(key, default)

    def configure_agent(self, agent) -> None:
        """Push the actionable keys into a live Agent."""
        for port in self.values.get("custom_protocol_ports", []):
            agent.add_custom_protocol_port(int(port))
