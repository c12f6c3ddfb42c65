"""Persistent JSON state store for restart survival
(ref pkg/state/store.go + types.go:9-321): subscribers, leases, pools,
sessions, NAT bindings — atomically written, loaded on start."""
from __future__ import annotations

import json
import os
import threading
import time
from typing import Dict, List, Optional

KINDS = ("subscribers", "leases", "pools", "sessions", "nat_bindings")


class StateStore:
    def __init__(self, path: str, autosave_interval: float = 0.0):
        self.path = path
        self._state: Dict[str, Dict[str, dict]] = {k: {} for k in KINDS}
        self._lock = threading.RLock()
        self._dirty = False
        self._stop = threading.Event()
        self._saver = None
        self.load()
        if autosave_interval > 0:
            self._saver = threading.Thread(
                target=self._save_loop, args=(autosave_interval,),
                daemon=True)
            self._saver.start()

    # --------------------------------------------------------------- CRUD
    def put(self, kind: str, key: str, obj: dict):
        with self._lock:
            self._state[kind][key] = dict(obj, _updated=time.time())
            self._dirty = True

    def get(self, kind: str, key: str) -> Optional[dict]:
        with self._lock:
            return self._state[kind].get(key)

    def delete(self, kind: str, key: str):
        with self._lock:
            self._state[kind].pop(key, None)
            self._dirty = True

    def list(self, kind: str) -> Dict[str, dict]:
        with self._lock:
            return dict(self._state[kind])

    # -------------------------------------------------------- persistence
    def save(self):
        with self._lock:
            if not self._dirty and os.path.exists(self.path):
                return
            blob = json.dumps(self._state)
            self._dirty = False
        tmp = self.path + ".tmp"
        with open(tmp, "w") as f:
            f.write(blob)
        os.replace(tmp, self.path)

    def load(self):
        if not os.path.exists(self.path):
            return
        try:
            with open(self.path) as f:
                data = json.load(f)
        except Exception:
            return
        with self._lock:
            for k in KINDS:
                self._state[k] = data.get(k, {})

    def _save_loop(self, interval: float):
        while not self._stop.wait(interval):
            try:
                self.save()
            except Exception:
                pass

    def close(self):
        self._stop.set()
        self.save()
