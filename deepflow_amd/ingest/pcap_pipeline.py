"""Raw pcap / packet-sequence storage (reference: server/ingester/pcap and
the l4_packet table fed by agent packet-sequence blocks).

Stores per-flow raw packet batches (bounded ring) with standard pcap export
so operators can pull a flow's packets (the "packet" tab / deepflow-ctl
pcap use case)."""
from __future__ import annotations

import struct
import time
from collections import OrderedDict
from typing import Dict, List, Optional, Tuple

from ..utils.stats import Counter

PCAP_MAGIC = 0xA1B2C3D4
LINKTYPE_ETHERNET = 1


class PcapPipeline:
    def __init__(self, max_flows: int = 4096, max_packets_per_flow: int = 256,
                 counter: Optional[Counter] = None):
        self.max_flows = max_flows
        self.max_packets = max_packets_per_flow
        # flow_id -> [(ts_ns, frame_bytes)]
        self.flows: "OrderedDict[int, List[Tuple[int, bytes]]]" = OrderedDict()
        self.counter = counter or Counter("ingester.pcap")

    def add_packet(self, flow_id: int, ts_ns: int, frame: bytes) -> None:
        entry = self.flows.get(flow_id)
        if entry is None:
            if len(self.flows) >= self.max_flows:
                self.flows.popitem(last=False)  # drop oldest flow
            entry = []
            self.flows[flow_id] = entry
        if len(entry) < self.max_packets:
            entry.append((ts_ns, frame))
            self.counter.add("packets_in")
        else:
            self.counter.add("packets_dropped")

    def ingest_payload(self, payload: bytes) -> int:
        """Wire format: repeated [flow_id u64][ts_ns u64][len u16][frame]."""
        pos, n = 0, 0
        while pos + 18 <= len(payload):
            flow_id, ts_ns, ln = struct.unpack_from("<QQH", payload, pos)
            pos += 18
            frame = payload[pos:pos + ln]
            pos += ln
            self.add_packet(flow_id, ts_ns, frame)
            n += 1
        return n

    def export_pcap(self, flow_id: int) -> Optional[bytes]:
        pkts = self.flows.get(flow_id)
        if not pkts:
            return None
        out = [struct.pack("<IHHiIII", PCAP_MAGIC, 2, 4, 0, 0, 65535,
                           LINKTYPE_ETHERNET)]
        for ts_ns, frame in pkts:
            out.append(struct.pack("<IIII", ts_ns // 10**9,
                                   (ts_ns % 10**9) // 1000,
                                   len(frame), len(frame)))
            out.append(frame)
        return b"".join(out)

    def stats(self) -> Dict[str, int]:
        return {"flows": len(self.flows),
                "packets": sum(len(v) for v in self.flows.values())}
