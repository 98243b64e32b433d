"""K8s ApiWatcher against a fake apiserver: LIST + WATCH (JSON lines),
events fold into platform/KG updates (reference api_watcher.rs)."""
import json
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import pytest

from deepflow_amd.agent.k8s_watcher import (K8sApiWatcher,
                                            platform_entries_from_pods)


def _pod(name, ip, ns="default", node="node-1", uid=None):
    return {"metadata": {"name": name, "uid": uid or name,
                         "namespace": ns, "labels": {"app": name}},
            "spec": {"nodeName": node},
            "status": {"podIP": ip}}


class FakeApiServer:
    def __init__(self):
        self.watch_events = []          # events served on next watch
        self.pods = [_pod("web-1", "10.2.0.5"), _pod("db-1", "10.2.0.9")]
        outer = self

        class H(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def do_GET(self):
                if self.path.startswith("/api/v1/pods?watch=true"):
                    self.send_response(200)
                    self.send_header("Content-Type", "application/json")
                    self.end_headers()
                    deadline = time.time() + 5
                    sent = 0
                    while time.time() < deadline:
                        while sent < len(outer.watch_events):
                            ev = outer.watch_events[sent]
                            self.wfile.write(
                                (json.dumps(ev) + "\n").encode())
                            self.wfile.flush()
                            sent += 1
                        time.sleep(0.05)
                elif self.path.startswith("/api/v1/pods"):
                    body = json.dumps({
                        "items": outer.pods,
                        "metadata": {"resourceVersion": "17"}}).encode()
                    self.send_response(200)
                    self.send_header("Content-Length", str(len(body)))
                    self.end_headers()
                    self.wfile.write(body)
                else:
                    self.send_response(404)
                    self.end_headers()

        self.httpd = ThreadingHTTPServer(("127.0.0.1", 0), H)
        self.port = self.httpd.server_address[1]
        threading.Thread(target=self.httpd.serve_forever,
                         daemon=True).start()

    def close(self):
        self.httpd.shutdown()


@pytest.mark.timeout(60)
def test_k8s_watch_to_platform():
    api = FakeApiServer()
    events = []
    w = K8sApiWatcher(f"http://127.0.0.1:{api.port}", events.append,
                      epc_id=3)
    try:
        w.start()
        t0 = time.time()
        while len(events) < 2 and time.time() - t0 < 5:
            time.sleep(0.05)
        assert {e["name"] for e in events} == {"web-1", "db-1"}  # LIST
        # live watch: a new pod appears, one dies
        api.watch_events.append({"type": "ADDED",
                                 "object": _pod("api-1", "10.2.0.20")})
        api.watch_events.append({"type": "DELETED",
                                 "object": _pod("db-1", "10.2.0.9")})
        while len(events) < 4 and time.time() - t0 < 8:
            time.sleep(0.05)
        assert any(e["name"] == "api-1" for e in events)
        assert any(e["name"] == "db-1" and e["deleted"] for e in events)
        assert "api-1" in {p["name"] for p in w.pods.values()}
        assert "db-1" not in {p["name"] for p in w.pods.values()}
        # fold into controller platform shape + push through the real
        # controller -> KnowledgeGraph path
        entries, names = platform_entries_from_pods(events)
        assert (3, 0x0A020014) in entries          # api-1 ip
        from deepflow_amd.control import ControllerLite
        from deepflow_amd.store.kg import KnowledgeGraphTable
        kg = KnowledgeGraphTable(device="cpu")
        ctl = ControllerLite(kg=kg)
        ctl.update_platform(entries, names=names)
        info = kg.lookup(3, 0x0A020014)
        assert info.pod_id != 0
        assert ctl.lookup_name("pod_map", info.pod_id) == "api-1"
    finally:
        w.stop()
        api.close()
