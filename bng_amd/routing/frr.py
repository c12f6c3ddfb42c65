"""FRR integration layer (ref pkg/routing/bgp.go:554-578): all BGP/BFD
configuration flows through an FRRExecutor abstraction (vtysh -c command
batches in production, a fake recording executor in tests — the same
seam the reference uses, subscriber_routes.go:127-131)."""
from __future__ import annotations

import subprocess
import threading
from typing import List, Optional


class FRRError(Exception):
    pass


class FRRExecutor:
    """Interface: run a batch of vtysh config commands."""

    def run(self, commands: List[str]) -> str:
        raise NotImplementedError


class VtyshExecutor(FRRExecutor):
    """Real `vtysh -c` execution (requires FRR on the host)."""

    def __init__(self, vtysh_path: str = "vtysh", timeout: float = 10.0):
        self.vtysh_path = vtysh_path
        self.timeout = timeout

    def run(self, commands: List[str]) -> str:
        argv = [self.vtysh_path]
        for c in commands:
            argv += ["-c", c]
        try:
            out = subprocess.run(argv, capture_output=True, text=True,
                                 timeout=self.timeout)
        except (subprocess.TimeoutExpired, FileNotFoundError) as e:
            raise FRRError(str(e))
        if out.returncode != 0:
            raise FRRError(out.stderr.strip() or f"rc={out.returncode}")
        return out.stdout


class FakeExecutor(FRRExecutor):
    """Recording executor for tests (ref bgp_test.go fake vtysh)."""

    def __init__(self, fail: bool = False):
        self.batches: List[List[str]] = []
        self.fail = fail
        self._lock = threading.Lock()

    def run(self, commands: List[str]) -> str:
        with self._lock:
            self.batches.append(list(commands))
        if self.fail:
            raise FRRError("fake failure")
        return ""

    def all_commands(self) -> List[str]:
        with self._lock:
            return [c for b in self.batches for c in b]
