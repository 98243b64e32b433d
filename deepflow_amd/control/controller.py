"""Controller-lite: agent registry, config push, platform data, tagrecorder.

The thin MI355X-native control plane covering the reference controller's
roles that the data plane depends on:
  - trisolaris-lite: agent (vtap) registry + versioned config/platform sync
    (reference controller/trisolaris, grpc/synchronizer/service.go:57)
  - platform data: (epc, ip) -> resource inventory pushed into the GPU
    KnowledgeGraph tables (reference PlatformData / AnalyzerSync)
  - tagrecorder-lite: resource id -> name maps served to the querier
    (reference controller/tagrecorder ch_* dictionaries)
  - election: single-process leader flag (k8s election is a deployment
    concern; the API surface is kept)
  - monitor: agent liveness from sync timestamps + analyzer rebalance hook

Transport: HTTP/JSON on the server app (the reference speaks gRPC; the
sync *semantics* — version-gated push of config/platform/groups — are
preserved; gRPC framing is round-2 work).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from fastapi import Request

from ..store.kg import KgInfo, KnowledgeGraphTable


@dataclass
class AgentRecord:
    agent_id: int
    hostname: str = ""
    ip: str = ""
    group: str = "default"
    first_seen: float = 0.0
    last_sync: float = 0.0
    config_version: int = 0
    platform_version: int = 0
    exceptions: int = 0
    analyzer: int = 0   # which shard/GPU ingests this agent's streams


DEFAULT_AGENT_CONFIG = {
    "max_millicpus": 1000,
    "max_memory": 768,          # MiB
    "sync_interval": 10,
    "l7_log_collect_enabled": True,
    "l4_log_collect_enabled": True,
    "throttle_per_second": 50000,
    "l7_protocols": ["HTTP", "DNS", "Redis", "MySQL"],
}


class ControllerLite:
    def __init__(self, kg: Optional[KnowledgeGraphTable] = None,
                 n_analyzers: int = 1, event_sink=None):
        self.event_sink = event_sink
        self.agents: Dict[int, AgentRecord] = {}
        self.group_configs: Dict[str, dict] = {"default": dict(DEFAULT_AGENT_CONFIG)}
        self.config_version = 1
        self.platform_version = 1
        self.platform: Dict[Tuple[int, int], KgInfo] = {}
        self.kg = kg
        self.n_analyzers = n_analyzers
        self.is_leader = True
        # tagrecorder-lite: (map_name, id) -> display name
        self.name_maps: Dict[str, Dict[int, str]] = {
            "device_map": {}, "pod_map": {}, "l3_epc_map": {},
            "vtap_map": {}, "service_map": {}, "az_map": {},
        }
        # genesis: agent-reported process/socket inventory + GPID allocation
        # (reference controller/genesis + gpid semantics)
        self.genesis_inventory: Dict[int, dict] = {}
        self._gpid_by_key: Dict[Tuple[int, int], int] = {}
        self._next_gpid = 1
        # prometheus global label encoder (reference
        # controller/prometheus: persistent metric/label-name/value ids
        # served to ingesters via GetPrometheusLabelIDs; metadb-backed
        # there, checkpoint-persisted here)
        self.prom_ids: Dict[str, Dict[str, int]] = {
            "metric": {}, "label_name": {}, "label_value": {}}

    # ------------------------------------------------------------ sync
    def sync(self, agent_id: int, hostname: str = "", ip: str = "",
             config_version: int = 0, platform_version: int = 0,
             exceptions: int = 0) -> dict:
        """Agent sync call (trident.Synchronizer/Sync analog): registers the
        agent and returns config/platform payloads only when versions moved
        (the reference's version-gated push)."""
        now = time.time()
        rec = self.agents.get(agent_id)
        if rec is None:
            rec = AgentRecord(agent_id=agent_id, first_seen=now)
            rec.analyzer = agent_id % self.n_analyzers
            self.agents[agent_id] = rec
        rec.hostname = hostname or rec.hostname
        rec.ip = ip or rec.ip
        rec.last_sync = now
        rec.exceptions = exceptions
        resp: dict = {
            "status": "ok",
            "config_version": self.config_version,
            "platform_version": self.platform_version,
            "analyzer": rec.analyzer,
        }
        if config_version != self.config_version:
            resp["config"] = self.group_configs.get(rec.group,
                                                    DEFAULT_AGENT_CONFIG)
        if platform_version != self.platform_version:
            resp["platform"] = [
                {"epc": epc, "ip": ip_, **vars(info)}
                for (epc, ip_), info in self.platform.items()]
        rec.config_version = self.config_version
        rec.platform_version = self.platform_version
        return resp

    # ------------------------------------------------------------ admin
    def set_group_config(self, group: str, config: dict) -> None:
        base = dict(DEFAULT_AGENT_CONFIG)
        base.update(config)
        self.group_configs[group] = base
        self.config_version += 1

    def update_platform(self, entries: Dict[Tuple[int, int], KgInfo],
                        names: Optional[Dict[str, Dict[int, str]]] = None
                        ) -> None:
        """Cloud/genesis-style inventory update -> bump version, refresh the
        GPU KnowledgeGraph table and tagrecorder name maps."""
        new_keys = [k for k in entries if k not in self.platform]
        self.platform.update(entries)
        self.platform_version += 1
        if self.kg is not None:
            self.kg.update(entries)
        if self.event_sink is not None:
            for (epc, ip_) in new_keys:
                info = entries[(epc, ip_)]
                self.event_sink("create", "pod" if info.pod_id else "device",
                                info.pod_id or info.l3_device_id,
                                resource_name="", description=f"epc={epc}")
        if names:
            for m, d in names.items():
                self.name_maps.setdefault(m, {}).update(d)

    def lookup_name(self, map_name: str, ident: int) -> Optional[str]:
        return self.name_maps.get(map_name, {}).get(ident)

    # ------------------------------------------------------------ genesis
    def genesis_report(self, agent_id: int, processes: List[dict],
                       sockets: List[dict]) -> dict:
        """Store the agent's inventory; allocate stable GPIDs per
        (agent, pid)."""
        self.genesis_inventory[agent_id] = {
            "processes": processes, "sockets": sockets,
            "reported_at": time.time(),
        }
        gpids = {}
        for proc in processes:
            key = (agent_id, proc["pid"])
            gpid = self._gpid_by_key.get(key)
            if gpid is None:
                gpid = self._next_gpid
                self._next_gpid += 1
                self._gpid_by_key[key] = gpid
            gpids[proc["pid"]] = gpid
        return {"status": "ok", "gpids": gpids}

    def lookup_gpid(self, agent_id: int, pid: int) -> int:
        return self._gpid_by_key.get((agent_id, pid), 0)

    # ------------------------------------------------------- prometheus
    def alloc_prom_ids(self, batch: Dict[str, List[str]]
                       ) -> Dict[str, Dict[str, int]]:
        """GetPrometheusLabelIDs analog: allocate (or return) persistent
        ids for metric names / label names / label values. Ids never
        change once issued (they survive restarts via the checkpoint)."""
        out: Dict[str, Dict[str, int]] = {}
        for kind, strings in batch.items():
            table = self.prom_ids[kind]
            res = {}
            for s in strings:
                i = table.get(s)
                if i is None:
                    i = len(table) + 1      # 0 reserved
                    table[s] = i
                res[s] = i
            out[kind] = res
        return out

    # ------------------------------------------------------- persistence
    # (reference keeps this state in MySQL metadb; here the registry,
    # platform inventory, name maps and GPID allocations serialize into
    # the checkpoint manifest so agents re-sync to the same ids after a
    # controller restart)
    def state_dict(self) -> dict:
        from dataclasses import asdict
        return {
            "agents": {aid: asdict(rec) for aid, rec in self.agents.items()},
            "group_configs": {g: dict(c)
                              for g, c in self.group_configs.items()},
            "config_version": self.config_version,
            "platform_version": self.platform_version,
            "platform": {k: asdict(v) for k, v in self.platform.items()},
            "name_maps": {m: dict(d) for m, d in self.name_maps.items()},
            "genesis_inventory": dict(self.genesis_inventory),
            "gpid_by_key": dict(self._gpid_by_key),
            "next_gpid": self._next_gpid,
            "prom_ids": {k: dict(v) for k, v in self.prom_ids.items()},
        }

    def load_state_dict(self, state: dict) -> None:
        self.agents = {aid: AgentRecord(**rec)
                       for aid, rec in state["agents"].items()}
        self.group_configs = {g: dict(c)
                              for g, c in state["group_configs"].items()}
        self.config_version = state["config_version"]
        self.platform_version = state["platform_version"]
        self.platform = {tuple(k): KgInfo(**v)
                         for k, v in state["platform"].items()}
        if self.kg is not None and self.platform:
            self.kg.update(self.platform)
        # in-place: the query engine holds a live reference to name_maps
        for m, d in state["name_maps"].items():
            self.name_maps.setdefault(m, {}).update(d)
        self.genesis_inventory = dict(state["genesis_inventory"])
        self._gpid_by_key = {tuple(k): v
                             for k, v in state["gpid_by_key"].items()}
        self._next_gpid = state["next_gpid"]
        for k, v in state.get("prom_ids", {}).items():
            self.prom_ids[k].update(v)

    # ------------------------------------------------------------ monitor
    def agent_status(self, stale_after_s: float = 60.0) -> List[dict]:
        now = time.time()
        out = []
        for rec in sorted(self.agents.values(), key=lambda r: r.agent_id):
            out.append({
                "agent_id": rec.agent_id,
                "hostname": rec.hostname,
                "ip": rec.ip,
                "group": rec.group,
                "analyzer": rec.analyzer,
                "alive": now - rec.last_sync < stale_after_s,
                "last_sync_age_s": round(now - rec.last_sync, 1),
                "exceptions": rec.exceptions,
            })
        return out

    def rebalance(self, traffic: Optional[Dict[int, float]] = None
                  ) -> Dict[int, int]:
        """Re-assign agents across analyzers. With per-agent traffic
        rates (bytes/s from receiver status or agent sync), bins are
        packed greedily heaviest-first onto the least-loaded analyzer
        (reference monitor/vtap/rebalance.go traffic-weighted mode);
        without rates, round-robin."""
        assignment: Dict[int, int] = {}
        if traffic:
            load = [0.0] * self.n_analyzers
            order = sorted(self.agents,
                           key=lambda a: -traffic.get(a, 0.0))
            for agent_id in order:
                tgt = min(range(self.n_analyzers), key=lambda i: load[i])
                load[tgt] += traffic.get(agent_id, 0.0)
                self.agents[agent_id].analyzer = tgt
                assignment[agent_id] = tgt
            return assignment
        for i, agent_id in enumerate(sorted(self.agents)):
            self.agents[agent_id].analyzer = i % self.n_analyzers
            assignment[agent_id] = self.agents[agent_id].analyzer
        return assignment

    # ------------------------------------------------------------ http
    def register(self, app) -> None:
        @app.post("/v1/sync/")
        async def sync(request: Request):
            body = await request.json()
            return self.sync(
                agent_id=int(body.get("agent_id", 0)),
                hostname=body.get("hostname", ""),
                ip=body.get("ip", ""),
                config_version=int(body.get("config_version", 0)),
                platform_version=int(body.get("platform_version", 0)),
                exceptions=int(body.get("exceptions", 0)))

        @app.get("/v1/agents/")
        def agents():
            return self.agent_status()

        @app.post("/v1/agent-group-config/{group}")
        async def set_config(group: str, request: Request):
            self.set_group_config(group, await request.json())
            return {"status": "ok", "config_version": self.config_version}

        @app.get("/v1/agent-group-config/{group}")
        def get_config(group: str):
            return {"group": group,
                    "config": self.group_configs.get(
                        group, dict(DEFAULT_AGENT_CONFIG)),
                    "config_version": self.config_version}

        @app.post("/v1/prometheus/label-ids/")
        async def prom_label_ids(request: Request):
            body = await request.json()
            return self.alloc_prom_ids(
                {k: list(v) for k, v in body.items()
                 if k in ("metric", "label_name", "label_value")})

        @app.post("/v1/rebalance/")
        async def rebalance(request: Request):
            body = {}
            try:
                body = await request.json()
            except Exception:
                pass
            traffic = {int(k): float(v)
                       for k, v in (body.get("traffic") or {}).items()}
            return self.rebalance(traffic or None)

        @app.post("/v1/genesis/")
        async def genesis(request: Request):
            body = await request.json()
            return self.genesis_report(int(body.get("agent_id", 0)),
                                       body.get("processes", []),
                                       body.get("sockets", []))

        @app.get("/v1/genesis/")
        def genesis_all():
            return {str(aid): g for aid, g in
                    self.genesis_inventory.items()}

        @app.get("/v1/genesis/{agent_id}")
        def genesis_get(agent_id: int):
            return self.genesis_inventory.get(agent_id, {})

        # -------- CRUD: domains (cloud platforms), vtap groups, orgs
        # (reference controller http API: /v1/domains, /v1/vtap-groups,
        # /v1/orgs — backing its MySQL tables; here in-process state)
        if not hasattr(self, "domains"):
            self.domains = {}
        if not hasattr(self, "vtap_groups"):
            self.vtap_groups = {"default": {"name": "default",
                                            "agents": []}}
        if not hasattr(self, "orgs"):
            self.orgs = {1: {"org_id": 1, "name": "default"}}

        @app.get("/v1/domains/")
        def list_domains():
            return list(self.domains.values())

        @app.post("/v1/domains/")
        async def create_domain(request: Request):
            body = await request.json()
            name = body["name"]
            self.domains[name] = {"name": name,
                                  "type": body.get("type", "kubernetes"),
                                  "config": body.get("config", {})}
            return self.domains[name]

        @app.delete("/v1/domains/{name}")
        def delete_domain(name: str):
            return {"deleted": self.domains.pop(name, None) is not None}

        @app.get("/v1/vtap-groups/")
        def list_groups():
            return list(self.vtap_groups.values())

        @app.post("/v1/vtap-groups/")
        async def create_group(request: Request):
            body = await request.json()
            name = body["name"]
            self.vtap_groups[name] = {"name": name, "agents": []}
            return self.vtap_groups[name]

        @app.post("/v1/vtap-groups/{name}/agents/{agent_id}")
        def assign_agent(name: str, agent_id: int):
            g = self.vtap_groups.get(name)
            if g is None:
                return {"error": "no such group"}
            if agent_id not in g["agents"]:
                g["agents"].append(agent_id)
            rec = self.agents.get(agent_id)
            if rec is not None:
                rec.group = name
            return g

        @app.get("/v1/orgs/")
        def list_orgs():
            return list(self.orgs.values())

        @app.post("/v1/orgs/")
        async def create_org(request: Request):
            body = await request.json()
            oid = int(body["org_id"])
            self.orgs[oid] = {"org_id": oid,
                              "name": body.get("name", f"org-{oid}")}
            return self.orgs[oid]
