"""Leader election (lease-file takeover) + cloud poller diff-push tests."""
import ipaddress

from deepflow_amd.control import ControllerLite
from deepflow_amd.control.cloud import CloudPoller, k8s_snapshot_to_platform
from deepflow_amd.control.election import LeaderElector


def test_election_takeover(tmp_path):
    lease = str(tmp_path / "leader.lease")
    a = LeaderElector(lease, "server-a", ttl_s=5)
    b = LeaderElector(lease, "server-b", ttl_s=5)
    t = 1000.0
    assert a.campaign(t) is True
    assert b.campaign(t + 1) is False       # lease held by a
    assert b.leader() is None or b._read()["leader"] == "server-a"
    assert a.campaign(t + 2) is True        # renewal
    # a dies; lease expires; b takes over
    assert b.campaign(t + 20) is True
    assert a.campaign(t + 21) is False      # a follows the new leader
    b.resign()
    assert a.campaign(t + 22) is True       # immediate re-election


SNAP1 = {
    "cluster_id": 3, "cluster_name": "prod",
    "pods": [
        {"id": 11, "name": "web-0", "ip": "10.0.0.5", "namespace": "shop",
         "node_id": 2, "epc": 1},
        {"id": 12, "name": "db-0", "ip": "10.0.0.6", "namespace": "shop",
         "node_id": 2, "epc": 1},
    ],
    "nodes": [{"id": 2, "name": "node-2", "ip": "10.0.1.2", "epc": 1}],
    "services": [{"id": 5, "name": "web-svc", "cluster_ip": "10.96.0.10",
                  "epc": 1}],
}


def test_k8s_normalize():
    entries, names = k8s_snapshot_to_platform(SNAP1)
    key = (1, int(ipaddress.IPv4Address("10.0.0.5")))
    assert entries[key].pod_id == 11
    assert entries[key].pod_cluster_id == 3
    assert names["pod"][11] == "web-0"
    assert names["pod_ns"][entries[key].pod_ns_id] == "shop"
    assert names["service"][5] == "web-svc"
    svc_key = (1, int(ipaddress.IPv4Address("10.96.0.10")))
    assert entries[svc_key].service_id == 5


def test_cloud_poller_diff_push():
    ctl = ControllerLite()
    snap = {"holder": dict(SNAP1)}
    poller = CloudPoller(ctl, source=lambda: snap["holder"])
    v0 = ctl.platform_version
    assert poller.poll_once() is True
    assert ctl.platform_version == v0 + 1
    assert poller.poll_once() is False      # unchanged snapshot: no push
    assert ctl.platform_version == v0 + 1
    snap2 = dict(SNAP1)
    snap2["pods"] = SNAP1["pods"] + [
        {"id": 13, "name": "web-1", "ip": "10.0.0.7", "namespace": "shop",
         "node_id": 2, "epc": 1}]
    snap["holder"] = snap2
    assert poller.poll_once() is True       # pod added -> version bump
    assert ctl.platform_version == v0 + 2
    assert poller.polls == 3 and poller.pushes == 2
    # tagrecorder name maps landed
    assert ctl.name_maps["pod"][13] == "web-1"


def test_pod_name_hydration_and_filter():
    """KG ids hydrate to display names in SQL results (dictGet analog),
    and name literals compile back to id filters."""
    from deepflow_amd.server import DeepflowServer
    from deepflow_amd.wire import pb, flow_log, framing
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12)
    poller = CloudPoller(srv.controller, source=lambda: SNAP1)
    poller.poll_once()

    def span(dst_ip, res):
        return {"base": {"start_time": 10**18, "end_time": 10**18 + 10**6,
                         "flow_id": 1, "vtap_id": 1, "tap_side": 1,
                         "head": {"proto": 20, "msg_type": 2, "rrt": 10},
                         "ip_src": 0x0A00000A,
                         "ip_dst": int(ipaddress.IPv4Address(dst_ip)),
                         "l3_epc_id_src": 1, "l3_epc_id_dst": 1,
                         "port_src": 999, "port_dst": 80, "protocol": 6},
                "req": {"req_type": "GET", "domain": "d", "resource": res,
                        "endpoint": res},
                "resp": {"status": 0, "code": 200}}

    spans = [span("10.0.0.5", "/to-web")] * 3 + [span("10.0.0.6", "/to-db")]
    srv.receiver.handle_frame(framing.encode_frame(
        framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG),
        framing.pack_records([pb.encode(s, flow_log.APP_PROTO_LOGS_DATA)
                              for s in spans])))
    r = srv.engine.query(
        "SELECT pod_name_1, Count(*) AS c FROM l7_flow_log "
        "GROUP BY pod_name_1 ORDER BY c DESC")
    assert r["values"] == [["web-0", 3], ["db-0", 1]]
    # filter by display name -> compiled to the pod id
    r2 = srv.engine.query(
        "SELECT Count(*) AS c FROM l7_flow_log WHERE pod_name_1 = 'db-0'")
    assert r2["values"] == [[1]]
    r3 = srv.engine.query(
        "SELECT Count(*) AS c FROM l7_flow_log "
        "WHERE pod_name_1 = 'no-such-pod'")
    assert r3["values"] in ([[0]], [])


def test_filereader_cloud_provider(tmp_path):
    """The filereader provider (reference controller/cloud/filereader):
    operator-maintained inventory file -> snapshot diff -> platform/KG."""
    import yaml
    from deepflow_amd.control import ControllerLite
    from deepflow_amd.control.cloud import FileReaderProvider
    from deepflow_amd.store.kg import KnowledgeGraphTable
    inv = tmp_path / "inventory.yaml"
    inv.write_text(yaml.safe_dump({
        "cluster_id": 5, "cluster_name": "edge-1",
        "pods": [{"name": "web-0", "ip": "10.9.0.4", "epc": 2, "id": 31}],
        "hosts": [{"name": "bm-01.dc", "ip": "10.9.1.1", "epc": 2,
                   "host_id": 7, "az": "az-east-1"}],
    }))
    kg = KnowledgeGraphTable(device="cpu")
    ctl = ControllerLite(kg=kg)
    prov = FileReaderProvider(ctl, str(inv), interval_s=30)
    assert prov.poll_once()
    assert kg.lookup(2, 0x0A090004).pod_id == 31
    assert kg.lookup(2, 0x0A090101).host_id == 7
    assert ctl.lookup_name("host", 7) == "bm-01.dc"
    # unchanged file -> no push; edited file -> diff push
    assert not prov.poll_once()
    data = yaml.safe_load(inv.read_text())
    data["hosts"].append({"name": "bm-02.dc", "ip": "10.9.1.2", "epc": 2,
                          "host_id": 8, "az": "az-east-1"})
    inv.write_text(yaml.safe_dump(data))
    assert prov.poll_once()
    assert kg.lookup(2, 0x0A090102).host_id == 8
