"""HA coordinator: DB-lease leader election
(reference: gpustack/server/coordinator/base.py:94 + election loop
server/server.py:1337-1379 — acquire/renew a TTL lease; leader-only tasks
run on the holder; lost leadership stops them).

Works over any shared database (SQLite file for dev, PostgreSQL/MySQL in
production) — no extra infrastructure, matching the DB-as-durable-state
design."""
from __future__ import annotations

import logging
import threading
import time
import uuid

from sqlalchemy import Column, Float, String

from ..db import Base, get_session

logger = logging.getLogger(__name__)


class Lease(Base):
    __tablename__ = "leases"
    name = Column(String(64), primary_key=True)
    holder = Column(String(64), nullable=False)
    expires_at = Column(Float, nullable=False)


class LeaseCoordinator:
    def __init__(self, name: str = "leader", ttl: float = 15.0,
                 holder: str | None = None, on_lost=None):
        self.name = name
        self.ttl = ttl
        self.holder = holder or uuid.uuid4().hex[:16]
        self.on_lost = on_lost
        self._leader = False
        self._stop = False

    @property
    def is_leader(self) -> bool:
        return self._leader

    def try_acquire(self) -> bool:
        now = time.time()
        try:
            with get_session() as s:
                row = s.get(Lease, self.name)
                if row is None:
                    s.add(Lease(name=self.name, holder=self.holder,
                                expires_at=now + self.ttl))
                    s.commit()
                    was = self._leader
                    self._leader = True
                    if not was:
                        logger.info("acquired leadership (%s)", self.holder)
                    return True
                if row.holder == self.holder or row.expires_at < now:
                    taking_over = row.holder != self.holder
                    row.holder = self.holder
                    row.expires_at = now + self.ttl
                    s.commit()
                    was = self._leader
                    self._leader = True
                    if not was or taking_over:
                        logger.info("acquired leadership (%s)%s", self.holder,
                                    " [takeover]" if taking_over else "")
                    return True
        except Exception:  # noqa: BLE001
            logger.exception("lease acquire failed")
        if self._leader:
            logger.warning("lost leadership (%s)", self.holder)
            self._leader = False
            if self.on_lost:
                self.on_lost()
        return False

    def run(self) -> None:
        """Renew loop (renew at ttl/3 cadence)."""
        while not self._stop:
            self.try_acquire()
            for _ in range(max(1, int(self.ttl / 3))):
                if self._stop:
                    return
                time.sleep(1.0)

    def start(self) -> threading.Thread:
        t = threading.Thread(target=self.run, name="coordinator", daemon=True)
        t.start()
        return t

    def stop(self) -> None:
        self._stop = True

    def release(self) -> None:
        self._stop = True
        try:
            with get_session() as s:
                row = s.get(Lease, self.name)
                if row is not None and row.holder == self.holder:
                    s.delete(row)
                    s.commit()
        except Exception:  # noqa: BLE001
            pass
        self._leader = False


class LocalCoordinator:
    """Single-node no-op coordinator (reference: coordinator/local.py:17)."""

    is_leader = True

    def start(self):
        return None

    def stop(self):
        pass

    def release(self):
        pass
