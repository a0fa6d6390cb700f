"""TTL cache with stampede protection and cross-instance invalidation.

Reference: gpustack server/cache.py:29-190 — a TTL cache with a
`locked_cached` decorator and coordinator-pub/sub invalidation. Our HA
layer is DB-lease based (server/coordinator.py) with no pub/sub channel,
so cross-instance invalidation is re-designed as a *generation counter*
in the shared DB: `invalidate()` bumps the row; readers re-check the
generation at most once per `check_interval` seconds, so steady-state
cache hits cost zero DB queries and a remote invalidation propagates
within `check_interval`.
"""
from __future__ import annotations

import functools
import threading
import time
from collections import OrderedDict

from sqlalchemy import Column, Integer, String

from ..db import Base, get_session


class CacheGeneration(Base):
    __tablename__ = "cache_generations"
    name = Column(String(128), primary_key=True)
    generation = Column(Integer, default=0, nullable=False)


def bump_generation(name: str) -> int:
    """Invalidate `name` across every server instance sharing the DB."""
    with get_session() as s:
        row = s.get(CacheGeneration, name)
        if row is None:
            row = CacheGeneration(name=name, generation=1)
            s.add(row)
        else:
            row.generation += 1
        s.commit()
        return row.generation


def read_generation(name: str) -> int:
    with get_session() as s:
        row = s.get(CacheGeneration, name)
        return row.generation if row else 0


class TTLCache:
    """Thread-safe TTL + LRU cache with optional distributed invalidation.

    When `distributed_name` is set, entries also carry the DB generation
    they were filled at; a newer generation (checked lazily, at most once
    per `check_interval`) invalidates every local entry.
    """

    def __init__(self, ttl: float = 60.0, maxsize: int = 1024,
                 distributed_name: str | None = None,
                 check_interval: float = 2.0):
        self.ttl = ttl
        self.maxsize = maxsize
        self.distributed_name = distributed_name
        self.check_interval = check_interval
        self._data: OrderedDict = OrderedDict()  # key -> (expires, gen, value)
        self._lock = threading.Lock()
        self._gen = 0
        self._last_check = 0.0

    def _current_gen(self) -> int:
        if self.distributed_name is None:
            return self._gen
        now = time.monotonic()
        if now - self._last_check >= self.check_interval:
            self._last_check = now
            try:
                self._gen = read_generation(self.distributed_name)
            except Exception:  # noqa: BLE001 — DB down: keep serving cached
                pass
        return self._gen

    def get(self, key, default=None):
        gen = self._current_gen()
        with self._lock:
            item = self._data.get(key)
            if item is None:
                return default
            expires, g, value = item
            if time.monotonic() >= expires or g < gen:
                del self._data[key]
                return default
            self._data.move_to_end(key)
            return value

    def set(self, key, value) -> None:
        gen = self._current_gen()
        with self._lock:
            self._data[key] = (time.monotonic() + self.ttl, gen, value)
            self._data.move_to_end(key)
            while len(self._data) > self.maxsize:
                self._data.popitem(last=False)

    def invalidate(self, key=None) -> None:
        """Drop one key (or all) locally; bump the DB generation so every
        other server instance drops its copies too."""
        with self._lock:
            if key is None:
                self._data.clear()
            else:
                self._data.pop(key, None)
        if self.distributed_name is not None:
            try:
                self._gen = bump_generation(self.distributed_name)
                self._last_check = time.monotonic()
            except Exception:  # noqa: BLE001
                pass
        else:
            self._gen += 1

    def __len__(self) -> int:
        with self._lock:
            return len(self._data)


_SENTINEL = object()


def locked_cached(cache: TTLCache, key_fn=None):
    """Memoize through `cache` with per-key locks: concurrent callers of a
    cold key compute once, the rest wait (stampede protection — reference
    server/cache.py `locked_cached`)."""
    key_locks: dict = {}
    meta_lock = threading.Lock()

    def deco(fn):
        @functools.wraps(fn)
        def wrapper(*args, **kwargs):
            key = key_fn(*args, **kwargs) if key_fn else (args, tuple(sorted(kwargs.items())))
            hit = cache.get(key, _SENTINEL)
            if hit is not _SENTINEL:
                return hit
            with meta_lock:
                lock = key_locks.setdefault(key, threading.Lock())
            with lock:
                hit = cache.get(key, _SENTINEL)  # filled while we waited?
                if hit is not _SENTINEL:
                    return hit
                value = fn(*args, **kwargs)
                cache.set(key, value)
                return value

        wrapper.cache = cache
        return wrapper

    return deco
