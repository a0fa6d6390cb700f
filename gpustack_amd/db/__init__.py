"""DB layer: SQLAlchemy ORM + active-record helpers + post-commit event bus.

Mirrors the reference's backbone (gpustack/mixins/active_record.py:95,777):
every create/update/delete publishes a CREATED/UPDATED/DELETED event on the
in-process bus after commit; controllers, the scheduler and watch streams
subscribe. Sync SQLAlchemy (FastAPI runs sync handlers in its threadpool);
SQLite by default, PostgreSQL/MySQL via database_url.
"""
from __future__ import annotations

import enum
import queue
import threading
import time
from contextlib import contextmanager
from typing import Any, Callable, Iterator

from sqlalchemy import create_engine, event as sa_event
from sqlalchemy.orm import DeclarativeBase, Session, sessionmaker


class Base(DeclarativeBase):
    pass


class EventType(str, enum.Enum):
    CREATED = "CREATED"
    UPDATED = "UPDATED"
    DELETED = "DELETED"
    HEARTBEAT = "HEARTBEAT"


class Event:
    __slots__ = ("type", "table", "data", "ts")

    def __init__(self, type: EventType, table: str, data: dict):
        self.type = type
        self.table = table
        self.data = data
        self.ts = time.time()

    def __repr__(self):
        return f"Event({self.type.value}, {self.table}, id={self.data.get('id')})"


class EventBus:
    """Thread-safe topic bus with bounded per-subscriber queues
    (reference: gpustack/server/bus.py:53-130)."""

    MAX_QUEUE = 2048

    def __init__(self):
        self._lock = threading.Lock()
        self._subs: dict[str, list[queue.Queue]] = {}
        self.dropped = 0

    def subscribe(self, table: str) -> queue.Queue:
        q: queue.Queue = queue.Queue(self.MAX_QUEUE)
        with self._lock:
            self._subs.setdefault(table, []).append(q)
        return q

    def unsubscribe(self, table: str, q: queue.Queue) -> None:
        with self._lock:
            subs = self._subs.get(table, [])
            if q in subs:
                subs.remove(q)

    def publish(self, ev: Event) -> None:
        with self._lock:
            subs = list(self._subs.get(ev.table, [])) + list(self._subs.get("*", []))
        for q in subs:
            try:
                q.put_nowait(ev)
            except queue.Full:
                self.dropped += 1


bus = EventBus()

_engine = None
_SessionLocal: sessionmaker | None = None


def init_db(database_url: str) -> None:
    global _engine, _SessionLocal
    kwargs: dict[str, Any] = {}
    if database_url.startswith("sqlite"):
        kwargs["connect_args"] = {"check_same_thread": False, "timeout": 30}
    _engine = create_engine(database_url, **kwargs)
    if database_url.startswith("sqlite"):
        @sa_event.listens_for(_engine, "connect")
        def _set_pragma(dbapi_conn, _):
            cur = dbapi_conn.cursor()
            cur.execute("PRAGMA journal_mode=WAL")
            cur.execute("PRAGMA foreign_keys=ON")
            cur.close()
    _SessionLocal = sessionmaker(bind=_engine, expire_on_commit=False)
    from ..schemas import tables  # noqa: F401  (register models)

    from sqlalchemy import inspect as _inspect

    fresh = not _inspect(_engine).get_table_names()
    Base.metadata.create_all(_engine)
    from . import migrations

    if fresh:
        migrations.stamp_head(_engine)   # new DB is already at head
    else:
        migrations.migrate(_engine)      # DB from an older build: upgrade


def get_engine():
    return _engine


@contextmanager
def session_scope() -> Iterator[Session]:
    assert _SessionLocal is not None, "init_db() not called"
    s = _SessionLocal()
    try:
        yield s
        s.commit()
    except Exception:
        s.rollback()
        raise
    finally:
        s.close()


def get_session() -> Session:
    assert _SessionLocal is not None, "init_db() not called"
    return _SessionLocal()


# -- active-record helpers (publish after commit) --------------------------

def ar_create(s: Session, obj) -> Any:
    s.add(obj)
    s.commit()
    s.refresh(obj)
    bus.publish(Event(EventType.CREATED, obj.__tablename__, obj.to_dict()))
    return obj


def ar_update(s: Session, obj) -> Any:
    s.add(obj)
    s.commit()
    s.refresh(obj)
    bus.publish(Event(EventType.UPDATED, obj.__tablename__, obj.to_dict()))
    return obj


def ar_delete(s: Session, obj) -> None:
    data = obj.to_dict()
    table = obj.__tablename__
    s.delete(obj)
    s.commit()
    bus.publish(Event(EventType.DELETED, table, data))


def watch_events(table: str, stop: Callable[[], bool] | None = None,
                 heartbeat: float = 15.0) -> Iterator[Event]:
    """Generator of events for a table, with HEARTBEAT keep-alives
    (backbone of the server->worker watch streams,
    reference: mixins/active_record.py:840)."""
    q = bus.subscribe(table)
    try:
        while True:
            if stop and stop():
                return
            try:
                yield q.get(timeout=heartbeat)
            except queue.Empty:
                yield Event(EventType.HEARTBEAT, table, {})
    finally:
        bus.unsubscribe(table, q)
