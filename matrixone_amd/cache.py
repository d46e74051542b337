"""Process-wide vector-index cache — mirror of pkg/vectorindex/cache/cache.go.

Semantics kept from the reference:
  - LoadOrStore keyed by index table key (cache.go:376-403 Search): a miss
    Load()s the implementation and stores it; concurrent misses load once.
  - TTL expiry (VectorIndexCacheTTL = 5 min, cache.go:60) swept by a
    housekeeping thread at TTL/2 cadence (cache.go:287-299).
  - Remove() destroys the entry; a search hitting a destroyed entry retries
    its load (cache.go:392-398).
Cross-CN staleness (StaleChecker PULL sweep) is the embedding host's concern;
the hook is kept (entry.is_stale callable) with the same eventual-consistency
contract (cache.go:95-105)."""

from __future__ import annotations

import threading
import time
from typing import Callable, Dict, Optional

VECTOR_INDEX_CACHE_TTL = 300.0  # cache.go:60 — 5 minutes


class _Entry:
    def __init__(self, impl):
        self.impl = impl
        self.last_used = time.monotonic()
        self.lock = threading.Lock()
        self.loaded = False
        self.destroyed = False
        self.is_stale: Optional[Callable[[], bool]] = None


class VectorIndexCache:
    def __init__(self, ttl: float = VECTOR_INDEX_CACHE_TTL):
        self.ttl = ttl
        self._entries: Dict[str, _Entry] = {}
        self._mu = threading.Lock()
        self._hk: Optional[threading.Thread] = None
        self._stop = threading.Event()

    def once(self):
        """cache.go:287 Once — start the housekeeping sweeper."""
        with self._mu:
            if self._hk is None:
                self._hk = threading.Thread(target=self._housekeep,
                                            daemon=True)
                self._hk.start()

    def _housekeep(self):
        tick = 0
        while not self._stop.wait(self.ttl / 2):
            tick += 1
            now = time.monotonic()
            with self._mu:
                keys = list(self._entries)
            for k in keys:
                e = self._entries.get(k)
                if e is None:
                    continue
                expired = now - e.last_used > self.ttl
                stale = False
                if tick % 4 == 0 and e.is_stale is not None:
                    try:
                        stale = e.is_stale()
                    except Exception:
                        stale = False  # transient error must not evict
                if expired or stale:
                    self.remove(k)

    def search(self, key: str, make_impl: Callable[[], object], proc, query,
               rt):
        """cache.go:376-403: LoadOrStore + load-once + retry on destroyed."""
        for _ in range(8):
            with self._mu:
                e = self._entries.get(key)
                if e is None:
                    e = _Entry(make_impl())
                    self._entries[key] = e
            with e.lock:
                if e.destroyed:
                    continue  # raced with Remove — retry with a fresh entry
                if not e.loaded:
                    e.impl.Load(proc)
                    e.loaded = True
                e.last_used = time.monotonic()
                return e.impl.Search(proc, query, rt)
        raise RuntimeError("vector index cache: persistent load/destroy race")

    def remove(self, key: str):
        with self._mu:
            e = self._entries.pop(key, None)
        if e is not None:
            with e.lock:
                e.destroyed = True
                if e.loaded:
                    e.impl.Destroy()
                    e.loaded = False

    def destroy(self):
        self._stop.set()
        with self._mu:
            keys = list(self._entries)
        for k in keys:
            self.remove(k)


Cache = VectorIndexCache()  # cache.go:62 process-wide instance
