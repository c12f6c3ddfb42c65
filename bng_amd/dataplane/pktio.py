"""Packet I/O — the NIC edge of the dataplane (AF_XDP
zero-copy rings feeding host-pinned batch buffers; ref pkg/ebpf attaches
XDP to the NIC driver, loader.go:294-315).

Sources/sinks are pluggable so the same pump drives:
  * SyntheticSource — generated traffic (bench, tests);
  * PcapSource / PcapSink — .pcap files (offline replay, test vectors);
  * AFPacketIO — Linux AF_PACKET raw socket (works everywhere Linux;
    the AF_XDP zero-copy upgrade binds the same interface with UMEM
    pinned for DMA — same Pump, different ring fill).

The Pump gathers frames into fixed-stride pinned batches, hands them to
the GPU launcher (uplink/dhcp pipelines), and routes results by verdict:
TX/FWD frames to the sink, PASS frames to the slow-path callback."""
from __future__ import annotations

import socket
import struct
import threading
import time
from typing import Callable, Iterable, List, Optional, Tuple

from . import abi

# --------------------------------------------------------------- pcap
PCAP_MAGIC = 0xA1B2C3D4


def pcap_write(path: str, frames: Iterable[bytes], linktype: int = 1):
    with open(path, "wb") as f:
        f.write(struct.pack("<IHHiIII", PCAP_MAGIC, 2, 4, 0, 0, 65535,
                            linktype))
        ts = 0
        for fr in frames:
            f.write(struct.pack("<IIII", ts, 0, len(fr), len(fr)))
            f.write(fr)
            ts += 1


def pcap_read(path: str) -> List[bytes]:
    out = []
    with open(path, "rb") as f:
        hdr = f.read(24)
        if len(hdr) < 24:
            return out
        magic = struct.unpack("<I", hdr[:4])[0]
        endian = "<" if magic == PCAP_MAGIC else ">"
        while True:
            ph = f.read(16)
            if len(ph) < 16:
                break
            _ts, _us, incl, _orig = struct.unpack(f"{endian}IIII", ph)
            out.append(f.read(incl))
    return out


class PcapSource:
    def __init__(self, path: str, loop: bool = False):
        self.frames = pcap_read(path)
        self.loop = loop
        self._i = 0

    def recv_batch(self, max_frames: int, timeout: float = 0.0) -> List[bytes]:
        out = []
        while len(out) < max_frames:
            if self._i >= len(self.frames):
                if not self.loop:
                    break
                self._i = 0
            out.append(self.frames[self._i])
            self._i += 1
        return out


class PcapSink:
    def __init__(self, path: str):
        self.path = path
        self.frames: List[bytes] = []

    def send_batch(self, frames: List[bytes]):
        self.frames.extend(frames)

    def close(self):
        pcap_write(self.path, self.frames)


def _concat_aranges(counts):
    """[3,1,2] -> [0,1,2, 0, 0,1] without a Python loop."""
    import numpy as np
    total = int(counts.sum())
    starts = np.cumsum(counts) - counts
    return np.arange(total, dtype=np.int64) - np.repeat(starts, counts)


def pack_frames(frames, stride: int):
    """Vectorized List[bytes] -> (data [n,stride] uint8, lens uint16).

    The ingest half of the host edge: one join + fancy-index scatter,
    no per-frame Python (round-1 VERDICT task 9)."""
    import numpy as np
    n = len(frames)
    lens = np.fromiter((len(f) for f in frames), dtype=np.int64, count=n)
    clip = np.minimum(lens, stride)
    joined = np.frombuffer(b"".join(frames), dtype=np.uint8)
    offs = np.cumsum(lens) - lens               # frame starts in joined
    data = np.zeros((n, stride), dtype=np.uint8)
    within = _concat_aranges(clip)
    dst = np.repeat(np.arange(n, dtype=np.int64) * stride, clip) + within
    src = np.repeat(offs, clip) + within
    data.reshape(-1)[dst] = joined[src]
    return data, clip.astype(np.uint16)


def unpack_frames(data, lens) -> List[bytes]:
    """[n,stride] + lens -> list of bytes (materialize only when a
    consumer genuinely needs Python objects)."""
    return [bytes(data[i, :lens[i]]) for i in range(len(lens))]


class ArraySink:
    """Array-native sink: accepts (data [m,stride] uint8, lens) batches
    without per-frame materialization.  AF_XDP TX rings and pcap writers
    consume this form directly; .frames materializes lazily for tests."""

    def __init__(self):
        self.batches: List[Tuple["object", "object"]] = []
        self.n = 0

    def send_batch_array(self, data, lens):
        self.batches.append((data.copy(), lens.copy()))
        self.n += len(lens)

    @property
    def frames(self) -> List[bytes]:
        return [f for d, l in self.batches for f in unpack_frames(d, l)]


class SyntheticSource:
    def __init__(self, generator: Callable[[int], List[bytes]]):
        self.generator = generator

    def recv_batch(self, max_frames: int, timeout: float = 0.0) -> List[bytes]:
        return self.generator(max_frames)


class ListSink:
    def __init__(self):
        self.frames: List[bytes] = []

    def send_batch(self, frames):
        self.frames.extend(frames)


class AFPacketIO:
    """Raw L2 I/O on a Linux interface (needs CAP_NET_RAW).  The AF_XDP
    zero-copy variant replaces recv/send with UMEM ring operations on
    the same interface; this socket path is the portable fallback the
    reference keeps for non-XDP drivers (loader.go generic mode)."""

    ETH_P_ALL = 0x0003

    def __init__(self, interface: str, mtu: int = 2048):
        self.interface = interface
        self.mtu = mtu
        self.sock = socket.socket(socket.AF_PACKET, socket.SOCK_RAW,
                                  socket.htons(self.ETH_P_ALL))
        self.sock.bind((interface, 0))
        self.sock.setblocking(False)

    def recv_batch(self, max_frames: int, timeout: float = 0.001) -> List[bytes]:
        import select
        out = []
        end = time.monotonic() + timeout
        while len(out) < max_frames:
            remain = end - time.monotonic()
            if remain <= 0:
                break
            r, _, _ = select.select([self.sock], [], [], remain)
            if not r:
                break
            try:
                while len(out) < max_frames:
                    out.append(self.sock.recv(self.mtu))
            except BlockingIOError:
                pass
        return out

    def send_batch(self, frames: List[bytes]):
        for fr in frames:
            try:
                self.sock.send(fr)
            except OSError:
                pass

    def close(self):
        self.sock.close()


class Pump:
    """Batch pump: source -> GPU pipeline -> sink/slow path.

    Gathers up to `batch` frames (or whatever arrived within
    `max_wait`), runs the launcher's uplink pipeline, then routes by
    verdict: TX (DHCP replies, truncated to out_len) and FWD (rewritten
    data packets) to the sink; PASS frames to the slow-path callback;
    DROP frames are counted and discarded."""

    def __init__(self, launcher, source, sink=None,
                 slow_path: Optional[Callable[[bytes], Optional[bytes]]] = None,
                 batch: int = 8192, stride: int = 512,
                 max_wait: float = 0.0005, sort_by_type: bool = True,
                 dhcp_service=None, direction: str = "uplink"):
        if direction not in ("uplink", "downlink"):
            raise ValueError("direction must be 'uplink' or 'downlink'")
        # "downlink" pumps the core-side NIC: NAT44 DNAT -> QoS egress
        # (ref tc_egress hook, tc.c) — return traffic toward subscribers
        self.direction = direction
        self.launcher = launcher
        self.source = source
        self.sink = sink
        self.slow_path = slow_path
        # optional persistent-kernel serving path: DHCP frames bypass
        # the batched kernel and go through the resident service's
        # doorbell (~31us p50 vs one batch residency)
        self.dhcp_service = dhcp_service
        self.batch = batch
        self.stride = stride
        self.max_wait = max_wait
        self.sort_by_type = sort_by_type
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.stats = {"batches": 0, "rx": 0, "tx": 0, "fwd": 0,
                      "passed": 0, "dropped": 0, "slow_replies": 0}

    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)

    def _loop(self):
        while not self._stop.is_set():
            if not self.pump_once():
                time.sleep(self.max_wait)

    def pump_once(self) -> int:
        frames = self.source.recv_batch(self.batch, self.max_wait)
        if not frames:
            return 0
        self.process(frames)
        return len(frames)

    def process(self, frames: List[bytes]) -> Tuple[List[bytes], List[bytes]]:
        """Run one batch; returns (out_frames, passed_frames)."""
        self.stats["batches"] += 1
        self.stats["rx"] += len(frames)
        if self.direction == "downlink":
            return self._process_downlink(frames)
        is_gpu = hasattr(self.launcher, "make_batch")
        out_frames: List[bytes] = []
        passed: List[bytes] = []
        if is_gpu:
            import numpy as np
            import torch
            data_np, lens_np = pack_frames(frames, self.stride)
            svc_mask = None
            if self.dhcp_service is not None and len(frames):
                # host-side pre-class: UDP dst 67 on untagged IPv4
                svc_mask = ((lens_np >= 42) &
                            (data_np[:, 12] == 0x08) &
                            (data_np[:, 13] == 0x00) &
                            (data_np[:, 23] == 17) &
                            (data_np[:, 36] == 0) &
                            (data_np[:, 37] == 67))
                if not svc_mask.any():
                    svc_mask = None
            svc_frames = None
            if svc_mask is not None:
                rows = np.nonzero(~svc_mask)[0]
                srows = np.nonzero(svc_mask)[0]
                svc_frames = [frames[i] for i in srows]
                sv, sol, srep = self.dhcp_service.serve(
                    data_np[srows], lens_np[srows],
                    int(time.time()))
                sv = sv.copy()
                sol = sol.view(np.uint16).copy()
                srep = srep.copy()
                data_np, lens_np = data_np[rows], lens_np[rows]
                frames = [frames[i] for i in rows]
            data = torch.from_numpy(data_np).to(self.launcher.device)
            lens = torch.from_numpy(lens_np.view(np.int16)).to(
                self.launcher.device)
            verdict, out_len = self.launcher.uplink(
                data, lens, sort_by_type=self.sort_by_type)
            v = verdict.cpu().numpy()
            ol = out_len.cpu().numpy().view(np.uint16)
            host = data.cpu().numpy()
            # vectorized verdict partitioning — no per-frame Python
            tx = v == abi.TX
            fwd = v == abi.FWD
            pas = v == abi.PASS
            self.stats["tx"] += int(tx.sum())
            self.stats["fwd"] += int(fwd.sum())
            self.stats["passed"] += int(pas.sum())
            self.stats["dropped"] += int(
                len(frames) - tx.sum() - fwd.sum() - pas.sum())
            out_mask = tx | fwd
            out_lens = np.where(tx, ol, lens_np).astype(np.uint16)[out_mask]
            out_data = host[out_mask]
            # PASS frames go to the slow path (few: cache misses only)
            passed = [frames[i] for i in np.nonzero(pas)[0]]
            if svc_mask is not None:
                stx = sv == abi.TX
                self.stats["tx"] += int(stx.sum())
                self.stats["passed"] += int((sv == abi.PASS).sum())
                if stx.any():
                    out_data = np.concatenate([out_data, srep[stx]])
                    out_lens = np.concatenate([out_lens, sol[stx]])
                # service misses (unknown subscriber) -> slow path
                passed.extend(svc_frames[i] for i in
                              np.nonzero(sv == abi.PASS)[0])
            slow_replies: List[bytes] = []
            if self.slow_path is not None:
                for fr in passed:
                    try:
                        reply = self.slow_path(fr)
                    except Exception:
                        reply = None
                    if reply:
                        # a slow path may answer with several frames
                        # (PPPoE handshakes emit PADS + LCP together)
                        if isinstance(reply, (list, tuple)):
                            slow_replies.extend(reply)
                            self.stats["slow_replies"] += len(reply)
                        else:
                            slow_replies.append(reply)
                            self.stats["slow_replies"] += 1
            if self.sink is not None:
                if hasattr(self.sink, "send_batch_array"):
                    if len(out_lens):
                        self.sink.send_batch_array(out_data, out_lens)
                    if slow_replies:
                        sd, sl = pack_frames(slow_replies, self.stride)
                        self.sink.send_batch_array(sd, sl)
                    return out_data, passed
                out_frames = unpack_frames(out_data, out_lens)
                out_frames.extend(slow_replies)
                self.sink.send_batch(out_frames)
                return out_frames, passed
            out_frames = unpack_frames(out_data, out_lens)
            out_frames.extend(slow_replies)
            return out_frames, passed
        else:
            # golden-model launcher (CPU mode); route like the fused
            # uplink kernel: DHCP frames that miss go to the slow path,
            # not down the data pipeline
            import struct as _st
            for fr in frames:
                # PPPoE/ARP ethertypes -> slow path (see uplink
                # kernel note; ARP for the gateway is answered there)
                if len(fr) >= 14 and _st.unpack_from(">H", fr, 12)[0] \
                        in (0x8863, 0x8864, 0x0806):
                    passed.append(fr)
                    self.stats["passed"] += 1
                    continue
                is_dhcp = (len(fr) >= 38 and
                           _st.unpack_from(">H", fr, 12)[0] in
                           (0x0800, 0x8100, 0x88A8))
                fb = bytearray(fr)
                vd, L = self.launcher.dp.dhcp_fastpath(fb)
                if vd == abi.TX:
                    out_frames.append(bytes(fb[:L]))
                    self.stats["tx"] += 1
                    continue
                if is_dhcp and len(fr) >= 38 and fr[23] == 17 and \
                        _st.unpack_from(">H", fr, 36)[0] == 67:
                    passed.append(fr)
                    self.stats["passed"] += 1
                    continue
                vd = self.launcher.dp.antispoof(bytes(fr))
                if vd == abi.FWD:
                    fb = bytearray(fr)
                    vd = self.launcher.dp.nat44_egress(fb)
                    if vd == abi.FWD:
                        vd = self.launcher.dp.qos(bytes(fb), "ingress")
                if vd == abi.FWD:
                    out_frames.append(bytes(fb))
                    self.stats["fwd"] += 1
                elif vd == abi.PASS:
                    passed.append(fr)
                    self.stats["passed"] += 1
                else:
                    self.stats["dropped"] += 1
        # slow path handles PASSed frames (DHCP slow path etc.)
        if self.slow_path is not None:
            for fr in passed:
                try:
                    reply = self.slow_path(fr)
                except Exception:
                    reply = None
                if reply:
                    if isinstance(reply, (list, tuple)):
                        out_frames.extend(reply)
                        self.stats["slow_replies"] += len(reply)
                    else:
                        out_frames.append(reply)
                        self.stats["slow_replies"] += 1
        if self.sink is not None and out_frames:
            self.sink.send_batch(out_frames)
        return out_frames, passed

    def _process_downlink(self, frames: List[bytes]) -> Tuple[List[bytes], List[bytes]]:
        """Core-side batch: fused NAT44 DNAT -> QoS-egress.  The
        downlink kernel rewrites frames in place and returns a verdict
        per frame (FWD toward the subscriber, DROP on shaping).  No
        DHCP/ARP/PPPoE handling here — control traffic lives on the
        access side."""
        is_gpu = hasattr(self.launcher, "make_batch")
        out_frames: List[bytes] = []
        passed: List[bytes] = []
        if is_gpu:
            import numpy as np
            import torch
            data_np, lens_np = pack_frames(frames, self.stride)
            data = torch.from_numpy(data_np).to(self.launcher.device)
            lens = torch.from_numpy(lens_np.view(np.int16)).to(
                self.launcher.device)
            verdict = self.launcher.downlink(data, lens)
            v = verdict.cpu().numpy()
            host = data.cpu().numpy()
            fwd = v == abi.FWD
            pas = v == abi.PASS
            self.stats["fwd"] += int(fwd.sum())
            self.stats["passed"] += int(pas.sum())
            self.stats["dropped"] += int(len(frames) - fwd.sum() - pas.sum())
            out_mask = fwd
            out_data = host[out_mask]
            out_lens = lens_np[out_mask]
            passed = [frames[i] for i in np.nonzero(pas)[0]]
            if self.sink is not None:
                if hasattr(self.sink, "send_batch_array"):
                    if len(out_lens):
                        self.sink.send_batch_array(out_data, out_lens)
                    return out_data, passed
                out_frames = unpack_frames(out_data, out_lens)
                self.sink.send_batch(out_frames)
                return out_frames, passed
            return unpack_frames(out_data, out_lens), passed
        for fr in frames:
            fb = bytearray(fr)
            vd = self.launcher.dp.nat44_ingress(fb)
            if vd == abi.FWD:
                vd = self.launcher.dp.qos(bytes(fb), "egress")
            if vd == abi.FWD:
                out_frames.append(bytes(fb))
                self.stats["fwd"] += 1
            elif vd == abi.PASS:
                passed.append(fr)
                self.stats["passed"] += 1
            else:
                self.stats["dropped"] += 1
        if self.sink is not None and out_frames:
            self.sink.send_batch(out_frames)
        return out_frames, passed
