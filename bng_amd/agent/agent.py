"""Nexus agent (ref pkg/agent/agent.go:41-77, types.go:42-231): node
state machine bootstrap -> connected -> partitioned -> recovering,
config watcher over the store, heartbeat."""
from __future__ import annotations

import json
import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from ..nexus.store import Store, TypedStore

S_BOOTSTRAP = "bootstrap"
S_CONNECTED = "connected"
S_PARTITIONED = "partitioned"
S_RECOVERING = "recovering"


class Agent:
    def __init__(self, store: Store, node_id: str,
                 heartbeat_interval: float = 5.0,
                 partition_after: float = 15.0):
        self.store = store
        self.node_id = node_id
        self.heartbeat_interval = heartbeat_interval
        self.partition_after = partition_after
        self.state = S_BOOTSTRAP
        self.config: Dict = {}
        self.config_store = TypedStore(store, "nexus/device_configs")
        self._listeners: List[Callable[[str, str], None]] = []
        self._config_listeners: List[Callable[[Dict], None]] = []
        self._last_ok = 0.0
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._watch_cancel = None

    def on_state_change(self, cb):
        self._listeners.append(cb)

    def on_config_change(self, cb):
        self._config_listeners.append(cb)

    def start(self):
        cfg = self.config_store.get(self.node_id)
        if cfg:
            self.config = cfg
        self._watch_cancel = self.config_store.watch(self._on_cfg)
        self._transition(S_CONNECTED)
        self._last_ok = time.time()
        t = threading.Thread(target=self._hb_loop, daemon=True)
        t.start()
        self._threads.append(t)
        return self

    def stop(self):
        self._stop.set()
        if self._watch_cancel:
            self._watch_cancel()

    def _on_cfg(self, typ, key, obj):
        if key != self.node_id:
            return
        self.config = obj or {}
        for cb in self._config_listeners:
            try:
                cb(self.config)
            except Exception:
                pass

    def _transition(self, new: str):
        old, self.state = self.state, new
        if old != new:
            for cb in self._listeners:
                try:
                    cb(old, new)
                except Exception:
                    pass

    def heartbeat_once(self) -> bool:
        try:
            self.store.put(f"nexus/heartbeats/{self.node_id}",
                           json.dumps({"ts": time.time(),
                                       "state": self.state}).encode())
            self._last_ok = time.time()
            if self.state == S_PARTITIONED:
                self._transition(S_RECOVERING)
            elif self.state == S_RECOVERING:
                self._transition(S_CONNECTED)
            return True
        except Exception:
            if self.state in (S_CONNECTED, S_RECOVERING) and \
                    time.time() - self._last_ok > self.partition_after:
                self._transition(S_PARTITIONED)
            return False

    def _hb_loop(self):
        while not self._stop.wait(self.heartbeat_interval):
            self.heartbeat_once()
