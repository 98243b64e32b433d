

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
