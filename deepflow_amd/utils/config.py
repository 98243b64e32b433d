"""Server configuration (reference: single /etc/server.yaml with per-module
subtrees re-parsed by each module, server/ingester/ingester.go:70-136).

Each module reads its own subtree; unknown keys are preserved (forward
compatibility, like the reference's yaml.Unmarshal into module structs).
"""
from __future__ import annotations

import copy
import os
from typing import Any, Dict, Optional

import yaml

DEFAULTS: Dict[str, Any] = {
    "ingester": {
        "listen-port": 20033,
        "segment-rows": 1 << 22,
        "dict-capacity": 1 << 22,
        "window-bytes": 200 << 30,   # hot window per GPU (of 288 GB HBM)
        "throttle-per-second": 0,    # 0 = unlimited
        "decode-queues": 2,
    },
    "querier": {
        "listen-port": 20416,
        "group-capacity": 1 << 20,
        "select-limit-default": 100,
    },
    "controller": {
        "sync-interval": 10,
        "agent-stale-seconds": 60,
        "analyzers": 1,
    },
    "profiler": {
        "gpu-capture": False,
        "flame-depth-limit": 128,
    },
    "self-telemetry": {
        "interval-seconds": 10,
        "enabled": True,
    },
}


class ServerConfig:
    def __init__(self, data: Optional[Dict[str, Any]] = None):
        self.data = copy.deepcopy(DEFAULTS)
        if data:
            self._merge(self.data, data)

    @staticmethod
    def _merge(base: Dict, over: Dict) -> None:
        for k, v in over.items():
            if isinstance(v, dict) and isinstance(base.get(k), dict):
                ServerConfig._merge(base[k], v)
            else:
                base[k] = v

    @classmethod
    def load(cls, path: Optional[str] = None) -> "ServerConfig":
        path = path or os.environ.get("DEEPFLOW_CONFIG", "/etc/server.yaml")
        data = None
        if path and os.path.exists(path):
            with open(path) as f:
                data = yaml.safe_load(f) or {}
        return cls(data)

    def module(self, name: str) -> Dict[str, Any]:
        return self.data.get(name, {})

    def get(self, module: str, key: str, default: Any = None) -> Any:
        return self.module(module).get(key, default)
