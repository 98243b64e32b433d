"""Robustness: the C++ flow engine must survive arbitrary garbage —
random payloads on every well-known port, truncated real frames, and
mutated golden captures — without crashing or reporting parse errors
for packets it chose to interpret."""
import os
import random
import struct

from deepflow_amd.agent import Agent
from deepflow_amd.agent.packets import eth_ipv4_tcp, SYN, SYNACK, PSH_ACK

PORTS = [53, 80, 443, 3306, 5432, 6379, 9092, 27017, 1883, 5672, 11211,
         9000, 4222, 12200, 18993, 6650, 61616, 8080, 30490, 50051]


def test_random_payload_fuzz():
    rng = random.Random(1234)
    a = Agent(vtap_id=1)
    t = 10**9
    for i in range(2000):
        port = PORTS[i % len(PORTS)]
        sport = 40000 + i
        payload = bytes(rng.getrandbits(8)
                        for _ in range(rng.randrange(1, 300)))
        a.packet(eth_ipv4_tcp(0x0A000001, 0x0A000002, sport, port, PSH_ACK,
                              1, 1, payload), t + i * 10**6)
        # sometimes a garbage "response"
        if i % 3 == 0:
            resp = bytes(rng.getrandbits(8)
                         for _ in range(rng.randrange(1, 200)))
            a.packet(eth_ipv4_tcp(0x0A000002, 0x0A000001, port, sport,
                                  PSH_ACK, 1, 1 + len(payload), resp),
                     t + i * 10**6 + 10**5)
    a.tick(1 << 62)
    st = a.stats()
    assert st["packets"] >= 2000
    a.drain(1)
    a.drain(0)
    a.close()


def test_truncated_frames():
    """Every prefix of a valid ethernet frame must be handled."""
    a = Agent(vtap_id=1)
    frame = eth_ipv4_tcp(0x0A000001, 0x0A000002, 40000, 80, PSH_ACK, 1, 1,
                         b"GET / HTTP/1.1\r\nHost: x\r\n\r\n")
    for cut in range(len(frame)):
        a.packet(frame[:cut], 10**9)
    a.packet(frame, 10**9)
    a.tick(1 << 62)
    a.close()


def test_mutated_golden_pcaps():
    """Bit-flip mutations of real captures through every parser."""
    import pytest
    FIX = "/root/reference/agent/resources/test/flow_generator"
    if not os.path.isdir(FIX):
        pytest.skip("reference fixtures not mounted")
    from tests.test_ref_pcaps import read_pcap
    rng = random.Random(7)
    names = ["dns/a-and-ns.pcap", "http/grpc-unary.pcap",
             "redis/redis.pcap", "zmtp/zmtp_null.pcap",
             "pulsar/pulsar-producer.pcap",
             "openwire/openwire_loose_producer.pcap",
             "some_ip/some_ip.pcap", "kafka/00-produce-v2.pcap"]
    for name in names:
        pkts = read_pcap(f"{FIX}/{name}")
        for trial in range(6):
            a = Agent(vtap_id=1)
            for ts, frame in pkts:
                b = bytearray(frame)
                for _ in range(1 + len(b) // 40):
                    b[rng.randrange(len(b))] ^= 1 << rng.randrange(8)
                a.packet(bytes(b), ts)
            a.tick(1 << 62)
            a.drain(1)
            a.drain(0)
            a.close()
