"""DHCP load-test harness (ref test/load/dhcp_benchmark.go:1-618 + CLI
cmd/dhcp-loadtest): configurable concurrency/duration/RPS/unique-MACs/
renewal-ratio, warmup phase, P50/P95/P99, fast-vs-slow split at 1 ms,
and target validation (ref MeetsTargets :579: >=50k RPS, P99<10ms,
>=95% hit rate; MeetsFastPathTarget :599: P99<100us)."""
from __future__ import annotations

import argparse
import concurrent.futures as cf
import json
import random
import time
from dataclasses import dataclass, field
from typing import Callable, List, Optional

FAST_PATH_THRESHOLD_S = 0.001      # 1 ms split (ref :481-508)


@dataclass
class Targets:
    min_rps: float = 50_000.0
    max_p99_s: float = 0.010
    min_hit_rate: float = 0.95
    fastpath_max_p99_s: float = 0.0001


@dataclass
class Result:
    total: int = 0
    errors: int = 0
    duration_s: float = 0.0
    latencies_s: List[float] = field(default_factory=list)

    def _pct(self, p: float) -> float:
        if not self.latencies_s:
            return 0.0
        xs = sorted(self.latencies_s)
        return xs[min(len(xs) - 1, int(len(xs) * p))]

    @property
    def rps(self) -> float:
        return self.total / self.duration_s if self.duration_s else 0.0

    @property
    def p50(self) -> float:
        return self._pct(0.50)

    @property
    def p95(self) -> float:
        return self._pct(0.95)

    @property
    def p99(self) -> float:
        return self._pct(0.99)

    @property
    def fast_count(self) -> int:
        return sum(1 for x in self.latencies_s
                   if x < FAST_PATH_THRESHOLD_S)

    @property
    def hit_rate(self) -> float:
        return self.fast_count / len(self.latencies_s) \
            if self.latencies_s else 0.0

    def fastpath_p99(self) -> float:
        xs = sorted(x for x in self.latencies_s
                    if x < FAST_PATH_THRESHOLD_S)
        return xs[min(len(xs) - 1, int(len(xs) * 0.99))] if xs else 0.0

    def meets_targets(self, t: Targets) -> List[str]:
        """ref MeetsTargets :579 — returns violated targets (empty=pass)."""
        bad = []
        if self.rps < t.min_rps:
            bad.append(f"rps {self.rps:.0f} < {t.min_rps:.0f}")
        if self.p99 > t.max_p99_s:
            bad.append(f"p99 {self.p99 * 1e3:.2f}ms > "
                       f"{t.max_p99_s * 1e3:.0f}ms")
        if self.hit_rate < t.min_hit_rate:
            bad.append(f"hit rate {self.hit_rate:.2%} < "
                       f"{t.min_hit_rate:.0%}")
        return bad

    def meets_fastpath_target(self, t: Targets) -> bool:
        """ref MeetsFastPathTarget :599."""
        return self.fastpath_p99() < t.fastpath_max_p99_s

    def report(self) -> dict:
        return {
            "total": self.total, "errors": self.errors,
            "duration_s": round(self.duration_s, 3),
            "rps": round(self.rps, 1),
            "p50_us": round(self.p50 * 1e6, 1),
            "p95_us": round(self.p95 * 1e6, 1),
            "p99_us": round(self.p99 * 1e6, 1),
            "fast_path_pct": round(self.hit_rate * 100, 2),
        }


class DHCPLoadTester:
    """Drives a handler(mac: bytes, renew: bool) -> bool; the handler is
    the in-process DHCP server (tests) or a UDP client (deployments)."""

    def __init__(self, handler: Callable[[bytes, bool], bool],
                 unique_macs: int = 1000, renewal_ratio: float = 0.8,
                 concurrency: int = 8, warmup: int = 100, seed: int = 1):
        self.handler = handler
        self.unique_macs = unique_macs
        self.renewal_ratio = renewal_ratio
        self.concurrency = concurrency
        self.warmup = warmup
        self.rng = random.Random(seed)

    def _mac(self, i: int) -> bytes:
        return (0xAA0000000000 + i).to_bytes(6, "big")

    def run(self, requests: int, duration_s: Optional[float] = None) -> Result:
        # warmup: establish leases so renewals hit the fast path
        for i in range(min(self.warmup, self.unique_macs)):
            try:
                self.handler(self._mac(i), False)
            except Exception:
                pass
        res = Result()
        deadline = time.perf_counter() + duration_s if duration_s else None

        def one(_):
            renew = self.rng.random() < self.renewal_ratio
            idx = self.rng.randrange(
                min(self.warmup, self.unique_macs) if renew
                else self.unique_macs)
            mac = self._mac(idx)
            t0 = time.perf_counter()
            try:
                ok = self.handler(mac, renew)
            except Exception:
                ok = False
            dt = time.perf_counter() - t0
            return ok, dt

        t_start = time.perf_counter()
        with cf.ThreadPoolExecutor(self.concurrency) as pool:
            done = 0
            while done < requests and \
                    (deadline is None or time.perf_counter() < deadline):
                batch = min(requests - done, self.concurrency * 8)
                for ok, dt in pool.map(one, range(batch)):
                    res.total += 1
                    res.latencies_s.append(dt)
                    if not ok:
                        res.errors += 1
                done += batch
        res.duration_s = time.perf_counter() - t_start
        return res


def main(argv=None) -> int:
    """CLI analog of cmd/dhcp-loadtest: -validate exits nonzero on
    missed targets."""
    ap = argparse.ArgumentParser(prog="dhcp-loadtest")
    ap.add_argument("--requests", type=int, default=20000)
    ap.add_argument("--concurrency", type=int, default=8)
    ap.add_argument("--unique-macs", type=int, default=1000)
    ap.add_argument("--renewal-ratio", type=float, default=0.8)
    ap.add_argument("--duration", type=float, default=0.0)
    ap.add_argument("--validate", action="store_true")
    ap.add_argument("--min-rps", type=float, default=50000)
    args = ap.parse_args(argv)

    # in-process target: the slow-path server with a golden fast path
    from ..dataplane.launcher import GoldenLauncher
    from ..dhcp import message as dm
    from ..dhcp.pool import PoolConfig, PoolManager
    from ..dhcp.server import DHCPServer
    launcher = GoldenLauncher()
    pm = PoolManager(launcher)
    pm.add_pool(PoolConfig(1, "10.0.0.0/16", gateway="10.0.0.1"))
    srv = DHCPServer(pm, "10.0.0.1")
    srv.set_launcher(launcher)

    def handler(mac, renew):
        mt = dm.REQUEST if renew else dm.DISCOVER
        return srv.handle(dm.build_request(mac, mt)) is not None

    tester = DHCPLoadTester(handler, args.unique_macs,
                            args.renewal_ratio, args.concurrency)
    res = tester.run(args.requests, args.duration or None)
    targets = Targets(min_rps=args.min_rps)
    report = res.report()
    report["violations"] = res.meets_targets(targets)
    print(json.dumps(report, indent=2))
    return 1 if (args.validate and report["violations"]) else 0


if __name__ == "__main__":
    import sys
    sys.exit(main())
