"""Custom L7 protocol plugins (reference: wasm + shared-object plugins,
agent/src/plugin/wasm/mod.rs check_payload/parse_payload surface).

The C++ core captures custom-protocol sessions generically (port rule ->
proto 127, raw request prefix in `resource`); plugins re-parse those records
before they leave the agent — the same parse_payload contract as the
reference's wasm VM, hosted in-process instead of a wasm runtime (wasmtime
is not in this image; the plugin ABI is the stable surface).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Callable, Dict, List, Optional

from ..wire import pb, flow_log, framing


@dataclass
class L7PluginInfo:
    req_type: str = ""
    domain: str = ""
    resource: str = ""
    endpoint: str = ""
    status: int = 0
    code: int = 0
    attributes: Optional[Dict[str, str]] = None


# parse_payload(raw_prefix: bytes, port: int) -> L7PluginInfo | None
ParseFn = Callable[[bytes, int], Optional[L7PluginInfo]]


class PluginHost:
    def __init__(self):
        self.plugins: List[ParseFn] = []
        self.parsed = 0
        self.unparsed = 0

    def register(self, parse: ParseFn) -> None:
        self.plugins.append(parse)

    def load_so(self, path: str) -> None:
        """dlopen a shared-object plugin (include/df_plugin.h ABI —
        the reference's plugin/shared_obj counterpart) and register it
        behind the same parse contract as python plugins."""
        import ctypes as ct

        class _Info(ct.Structure):
            _fields_ = [("req_type", ct.c_char * 32),
                        ("domain", ct.c_char * 128),
                        ("resource", ct.c_char * 256),
                        ("endpoint", ct.c_char * 128),
                        ("status", ct.c_int32), ("code", ct.c_int32)]

        lib = ct.CDLL(path)
        fn = lib.df_plugin_parse
        fn.restype = ct.c_int
        fn.argtypes = [ct.c_char_p, ct.c_uint32, ct.c_uint16,
                       ct.POINTER(_Info)]

        def parse(raw: bytes, port: int) -> Optional[L7PluginInfo]:
            info = _Info()
            if fn(raw, len(raw), port, ct.byref(info)) != 1:
                return None
            return L7PluginInfo(
                req_type=info.req_type.decode("utf-8", "replace"),
                domain=info.domain.decode("utf-8", "replace"),
                resource=info.resource.decode("utf-8", "replace"),
                endpoint=info.endpoint.decode("utf-8", "replace"),
                status=info.status, code=info.code)

        self.register(parse)

    def process_l7_payload(self, payload: bytes) -> bytes:
        """Rewrite custom-protocol (127) records via registered plugins;
        other records pass through untouched."""
        out = []
        for rec in framing.iter_records(payload):
            d = pb.decode(rec, flow_log.APP_PROTO_LOGS_DATA)
            if d.get("base", {}).get("head", {}).get("proto") != 127 or \
                    not self.plugins:
                out.append(rec)
                continue
            raw = d.get("req", {}).get("resource", "").encode(
                "utf-8", "surrogateescape")
            port = d.get("base", {}).get("port_dst", 0)
            info = None
            for parse in self.plugins:
                info = parse(raw, port)
                if info is not None:
                    break
            if info is None:
                self.unparsed += 1
                out.append(rec)
                continue
            d.setdefault("req", {})
            d["req"]["req_type"] = info.req_type
            d["req"]["domain"] = info.domain
            d["req"]["resource"] = info.resource
            d["req"]["endpoint"] = info.endpoint
            d.setdefault("resp", {})
            d["resp"]["status"] = info.status
            d["resp"]["code"] = info.code
            if info.attributes:
                ext = d.setdefault("ext_info", {})
                ext.setdefault("attribute_names", []).extend(
                    info.attributes.keys())
                ext.setdefault("attribute_values", []).extend(
                    info.attributes.values())
            self.parsed += 1
            out.append(pb.encode(d, flow_log.APP_PROTO_LOGS_DATA))
        return framing.pack_records(out)
