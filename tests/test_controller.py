

def test_crud_routes():
    """Domain / vtap-group / org CRUD (controller http API surface)."""
    from fastapi.testclient import TestClient
    from fastapi import FastAPI
    from deepflow_amd.control import ControllerLite
    ctl = ControllerLite()
    app = FastAPI()
    ctl.register(app)
    c = TestClient(app)
    assert c.post("/v1/domains/", json={"name": "prod-k8s",
                                        "type": "kubernetes"}).json()[
        "name"] == "prod-k8s"
    assert [d["name"] for d in c.get("/v1/domains/").json()] == ["prod-k8s"]
    assert c.delete("/v1/domains/prod-k8s").json()["deleted"] is True
    assert c.get("/v1/domains/").json() == []

    c.post("/v1/vtap-groups/", json={"name": "edge"})
    ctl.sync(agent_id=5)
    g = c.post("/v1/vtap-groups/edge/agents/5").json()
    assert g["agents"] == [5]
    assert ctl.agents[5].group == "edge"
    # group-scoped config now reaches agent 5
    c.post("/v1/agent-group-config/edge", json={"sync_interval": 5})
    resp = ctl.sync(agent_id=5, config_version=0)
    assert resp["config"]["sync_interval"] == 5

    c.post("/v1/orgs/", json={"org_id": 7, "name": "tenant-7"})
    assert {o["org_id"] for o in c.get("/v1/orgs/").json()} == {1, 7}


def test_alert_policies():
    """DF-SQL alert policy fires into alert_event, queryable back."""
    from deepflow_amd.gen import SpanGenConfig
    from deepflow_amd.gen.spans import gen_span_payload
    from deepflow_amd.ingest import L7IngestPipeline
    from deepflow_amd.ingest.event_pipeline import EventPipeline
    from deepflow_amd.query import QueryEngine
    from deepflow_amd.control.alerting import (AlertEvaluator, AlertPolicy,
                                               LEVEL_CRITICAL)
    cfg = SpanGenConfig(n=500, seed=11, tag_cardinality=20, n_ips=32,
                        n_services=3, n_resources=6)
    pipe = L7IngestPipeline(device="cpu", segment_rows=1 << 10,
                            dict_capacity=1 << 11,
                            time_base_s=cfg.base_time_ns // 10**9)
    pipe.ingest_frame_payload(gen_span_payload(cfg))
    eng = QueryEngine(pipe, device="cpu")
    events = EventPipeline()
    ev = AlertEvaluator(eng, events)
    ev.add_policy(AlertPolicy(
        "too-many-errors",
        "SELECT request_resource, Count(*) AS errs FROM l7_flow_log "
        "WHERE response_status = 'Server Error' GROUP BY request_resource",
        column="errs", op=">=", threshold=1, level=LEVEL_CRITICAL,
        target_column="request_resource"))
    ev.add_policy(AlertPolicy(
        "never-fires",
        "SELECT Count(*) AS c FROM l7_flow_log", column="c", op=">",
        threshold=10**9))
    fired = ev.evaluate_once()
    assert fired >= 1
    assert all(e["policy_name"] == "too-many-errors"
               for e in events.alert_events)
    assert events.alert_events[0]["level"] == LEVEL_CRITICAL
    assert events.alert_events[0]["target"].startswith("/")
    # and it is queryable through the alert_event table
    eng.alert_event_rows = lambda: events.alert_events
    r = eng.query("SELECT policy_name, level FROM alert_event LIMIT 5")
    assert ["too-many-errors", LEVEL_CRITICAL] in r["values"]


def test_traffic_weighted_rebalance():
    """Heavy agents spread across analyzers by load, not id order
    (reference monitor/vtap/rebalance.go)."""
    from deepflow_amd.control import ControllerLite
    c = ControllerLite(n_analyzers=2)
    for aid in range(1, 5):
        c.sync(aid)
    # agents 1+2 are heavy; round-robin would pair them on one analyzer
    traffic = {1: 1000.0, 2: 900.0, 3: 10.0, 4: 5.0}
    assign = c.rebalance(traffic)
    assert assign[1] != assign[2]          # heaviest two split
    loads = [sum(traffic[a] for a, t in assign.items() if t == i)
             for i in range(2)]
    assert abs(loads[0] - loads[1]) <= 100  # near-balanced
    # no traffic -> round-robin still works
    assign2 = c.rebalance()
    assert sorted(assign2.values()) == [0, 0, 1, 1]
