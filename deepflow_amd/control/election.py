"""Leader election + failover for multi-process server deployments.

The reference elects a master controller through its MySQL/etcd-backed
election (server/controller/election); on a single MI355X node the
natural primitive is an O_EXCL heartbeat lease file on shared disk:
the leader renews an (instance_id, deadline) lease; any candidate may
take over once the lease expires. Tie-breaks go through an atomic
rename (os.rename is atomic within a filesystem).
"""
from __future__ import annotations

import json
import os
import time
from typing import Optional


class LeaderElector:
    def __init__(self, lease_path: str, instance_id: str,
                 ttl_s: float = 5.0):
        self.lease_path = lease_path
        self.instance_id = instance_id
        self.ttl_s = ttl_s
        self.is_leader = False

    def _read(self) -> Optional[dict]:
        try:
            with open(self.lease_path) as f:
                return json.load(f)
        except (OSError, ValueError):
            return None

    def _write(self, now: float) -> bool:
        tmp = f"{self.lease_path}.{self.instance_id}.tmp"
        try:
            with open(tmp, "w") as f:
                json.dump({"leader": self.instance_id,
                           "deadline": now + self.ttl_s}, f)
            os.replace(tmp, self.lease_path)
            return True
        except OSError:
            return False

    def campaign(self, now: Optional[float] = None) -> bool:
        """One election tick: renew if leader, take over if the lease is
        free or expired, otherwise follow. Returns is_leader."""
        now = time.time() if now is None else now
        lease = self._read()
        if lease is None or lease.get("deadline", 0) < now or \
                lease.get("leader") == self.instance_id:
            if self._write(now):
                # re-read to resolve near-simultaneous takeovers: the
                # last atomic rename wins and everyone agrees on it
                lease = self._read()
                self.is_leader = bool(lease) and \
                    lease.get("leader") == self.instance_id
            else:
                self.is_leader = False
        else:
            self.is_leader = False
        return self.is_leader

    def resign(self) -> None:
        if self.is_leader:
            try:
                os.unlink(self.lease_path)
            except OSError:
                pass
            self.is_leader = False

    def leader(self) -> Optional[str]:
        lease = self._read()
        if lease and lease.get("deadline", 0) >= time.time():
            return lease.get("leader")
        return None
