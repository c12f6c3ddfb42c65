"""Nexus store abstraction — K/V + watch.

Python re-design of the reference's nexus.Store interface
(pkg/nexus/store.go:13-40) with MemoryStore (:43-126) and TypedStore
(:129-209).  Multi-node behavior is tested in-process by running several
components against one shared MemoryStore (SURVEY.md §4 layer 5).
"""
from __future__ import annotations

import json
import threading
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional


@dataclass
class WatchEvent:
    type: str          # "put" | "delete"
    key: str
    value: Optional[bytes] = None


class Store:
    """K/V + prefix list + watch (ref store.go:13-40)."""

    def get(self, key: str) -> Optional[bytes]:
        raise NotImplementedError

    def put(self, key: str, value: bytes) -> None:
        raise NotImplementedError

    def delete(self, key: str) -> None:
        raise NotImplementedError

    def list(self, prefix: str) -> Dict[str, bytes]:
        raise NotImplementedError

    def watch(self, prefix: str,
              callback: Callable[[WatchEvent], None]) -> Callable[[], None]:
        """Register a watcher; returns an unsubscribe function."""
        raise NotImplementedError

    def close(self) -> None:
        pass


class MemoryStore(Store):
    """Thread-safe in-memory Store (ref store.go:43-126); the in-process
    multi-node test substrate."""

    def __init__(self):
        self._data: Dict[str, bytes] = {}
        self._watchers: List[tuple] = []   # (prefix, callback)
        self._lock = threading.RLock()

    def get(self, key):
        with self._lock:
            return self._data.get(key)

    def put(self, key, value):
        if isinstance(value, str):
            value = value.encode()
        with self._lock:
            self._data[key] = bytes(value)
            watchers = [w for w in self._watchers if key.startswith(w[0])]
        for _, cb in watchers:
            cb(WatchEvent("put", key, bytes(value)))

    def delete(self, key):
        with self._lock:
            existed = self._data.pop(key, None)
            watchers = [w for w in self._watchers if key.startswith(w[0])]
        if existed is not None:
            for _, cb in watchers:
                cb(WatchEvent("delete", key))

    def list(self, prefix):
        with self._lock:
            return {k: v for k, v in self._data.items()
                    if k.startswith(prefix)}

    def watch(self, prefix, callback):
        ent = (prefix, callback)
        with self._lock:
            self._watchers.append(ent)

        def cancel():
            with self._lock:
                if ent in self._watchers:
                    self._watchers.remove(ent)
        return cancel


class TypedStore:
    """JSON-typed view over a Store (ref store.go TypedStore[T]:129-209)."""

    def __init__(self, store: Store, prefix: str):
        self.store = store
        self.prefix = prefix.rstrip("/") + "/"

    def _k(self, key: str) -> str:
        return self.prefix + key

    def get(self, key: str) -> Optional[dict]:
        raw = self.store.get(self._k(key))
        return None if raw is None else json.loads(raw)

    def put(self, key: str, obj: Any) -> None:
        self.store.put(self._k(key), json.dumps(obj).encode())

    def delete(self, key: str) -> None:
        self.store.delete(self._k(key))

    def list(self) -> Dict[str, dict]:
        out = {}
        for k, v in self.store.list(self.prefix).items():
            out[k[len(self.prefix):]] = json.loads(v)
        return out

    def watch(self, callback):
        def cb(ev: WatchEvent):
            obj = json.loads(ev.value) if ev.value else None
            callback(ev.type, ev.key[len(self.prefix):], obj)
        return self.store.watch(self.prefix, cb)
