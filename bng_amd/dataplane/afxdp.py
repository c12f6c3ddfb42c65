"""AF_XDP zero-copy-capable packet rings — the NIC edge of the MI355X
dataplane (round-1 VERDICT task 1; ref pkg/ebpf/loader.go:294-315
attaches XDP driver-mode with generic fallback).

Pure-ctypes implementation (no libbpf/libxdp in the image):

  * UMEM: one mmap'd frame arena registered with XDP_UMEM_REG.  For the
    GPU path the arena is allocated as a PINNED host buffer (torch
    pin_memory) so RX frames are directly DMA-able to the device with
    hipMemcpyAsync — the "UMEM pinned for DMA" design the survey names.
  * Four SPSC rings (fill/completion/RX/TX) mmap'd from the socket;
    x86-TSO store ordering + CPython's sequential bytecode give the
    producer/consumer protocol its required ordering on this host.
  * A 6-instruction XDP redirect program (rx_queue_index -> XSKMAP) is
    assembled inline, loaded with BPF_PROG_LOAD and attached with a
    BPF_LINK_CREATE XDP link — the same default-program shape libxdp
    installs.  Driver (zero-copy) mode is requested first; generic/SKB
    copy mode is the fallback (veth, lo, and other non-ZC drivers).

The socket exposes the Pump source/sink API (recv_batch/send_batch), so
`bng run --xdp-iface ethN` swaps it in for the AF_PACKET fallback with
no other changes.
"""
from __future__ import annotations

import ctypes
import mmap
import os
import select
import socket
import struct
from typing import List, Tuple

_libc = ctypes.CDLL(None, use_errno=True)
_SYS_BPF = 321            # x86_64

# bpf(2) commands
BPF_MAP_CREATE = 0
BPF_MAP_UPDATE_ELEM = 2
BPF_MAP_DELETE_ELEM = 3
BPF_PROG_LOAD = 5
BPF_LINK_CREATE = 28

BPF_MAP_TYPE_XSKMAP = 17
BPF_PROG_TYPE_XDP = 6
BPF_XDP = 37              # attach_type

XDP_FLAGS_SKB_MODE = 1 << 1
XDP_FLAGS_DRV_MODE = 1 << 2

# sockopt level/optnames (linux/if_xdp.h)
SOL_XDP = 283
XDP_MMAP_OFFSETS = 1
XDP_RX_RING = 2
XDP_TX_RING = 3
XDP_UMEM_REG = 4
XDP_UMEM_FILL_RING = 5
XDP_UMEM_COMPLETION_RING = 6
XDP_STATISTICS = 7

# bind flags
XDP_SHARED_UMEM = 1 << 0
XDP_COPY = 1 << 1
XDP_ZEROCOPY = 1 << 2
XDP_USE_NEED_WAKEUP = 1 << 3

# mmap pgoffsets
_PGOFF_RX = 0
_PGOFF_TX = 0x80000000
_PGOFF_FILL = 0x100000000
_PGOFF_COMP = 0x180000000

XDP_PASS = 2


def _bpf(cmd: int, attr: bytes) -> int:
    buf = ctypes.create_string_buffer(attr, len(attr))
    r = _libc.syscall(_SYS_BPF, cmd, buf, len(attr))
    if r < 0:
        e = ctypes.get_errno()
        raise OSError(e, f"bpf(cmd={cmd}): {os.strerror(e)}")
    return r


def _insn(op: int, dst: int, src: int, off: int, imm: int) -> bytes:
    return struct.pack("<BBhi", op, (src << 4) | dst, off, imm)


def load_redirect_prog(xskmap_fd: int) -> int:
    """XDP prog: return bpf_redirect_map(xskmap, ctx->rx_queue_index,
    XDP_PASS) — redirect every frame on the bound queue to the XSK,
    pass traffic on queues with no socket (libxdp's default shape)."""
    insns = (_insn(0x61, 2, 1, 16, 0) +                   # r2 = rx_queue_idx
             _insn(0x18, 1, 1, 0, xskmap_fd) +            # r1 = map (ld64)
             _insn(0x00, 0, 0, 0, 0) +
             _insn(0xb7, 3, 0, 0, XDP_PASS) +             # r3 = fallback
             _insn(0x85, 0, 0, 0, 51) +                   # redirect_map
             _insn(0x95, 0, 0, 0, 0))                     # exit
    ib = ctypes.create_string_buffer(insns, len(insns))
    lic = ctypes.create_string_buffer(b"GPL\0")
    attr = struct.pack("<IIQQIIQI", BPF_PROG_TYPE_XDP, len(insns) // 8,
                       ctypes.addressof(ib), ctypes.addressof(lic),
                       0, 0, 0, 0) + b"\x00" * 64
    return _bpf(BPF_PROG_LOAD, attr)


def create_xskmap(n_queues: int = 4) -> int:
    attr = struct.pack("<IIIII", BPF_MAP_TYPE_XSKMAP, 4, 4, n_queues, 0)
    return _bpf(BPF_MAP_CREATE, attr + b"\x00" * 100)


def xskmap_set(map_fd: int, queue: int, sock_fd: int):
    k = ctypes.c_uint32(queue)
    v = ctypes.c_uint32(sock_fd)
    attr = struct.pack("<IxxxxQQQ", map_fd, ctypes.addressof(k),
                       ctypes.addressof(v), 0)
    _bpf(BPF_MAP_UPDATE_ELEM, attr)


def attach_xdp(prog_fd: int, ifindex: int, mode_flags: int) -> int:
    """BPF_LINK_CREATE XDP link; closing the returned fd detaches."""
    attr = struct.pack("<IIII", prog_fd, ifindex, BPF_XDP,
                       mode_flags) + b"\x00" * 32
    return _bpf(BPF_LINK_CREATE, attr)


# ------------------------------------------------------------ rtnetlink
NLM_F_REQUEST, NLM_F_ACK = 1, 4
NLM_F_CREATE, NLM_F_EXCL = 0x400, 0x200
RTM_NEWLINK, RTM_DELLINK = 16, 17
IFLA_IFNAME, IFLA_MTU, IFLA_LINKINFO = 3, 4, 18
IFLA_INFO_KIND, IFLA_INFO_DATA = 1, 2
VETH_INFO_PEER = 1
IFF_UP = 1


def _nlattr(t: int, payload: bytes) -> bytes:
    ln = 4 + len(payload)
    return struct.pack("<HH", ln, t) + payload + b"\x00" * ((4 - ln % 4) % 4)


def _rtnl(msg_type: int, flags: int, payload: bytes):
    sk = socket.socket(socket.AF_NETLINK, socket.SOCK_RAW, 0)
    try:
        sk.bind((0, 0))
        hdr = struct.pack("<IHHII", 16 + len(payload), msg_type, flags, 1, 0)
        sk.send(hdr + payload)
        resp = sk.recv(65536)
    finally:
        sk.close()
    _, ty, *_ = struct.unpack("<IHHII", resp[:16])
    if ty == 2:                                  # NLMSG_ERROR (0 = ack)
        err = struct.unpack("<i", resp[16:20])[0]
        if err:
            raise OSError(-err, os.strerror(-err))


def _ifinfo(index: int = 0, flags: int = 0, change: int = 0) -> bytes:
    return struct.pack("<BxHiII", 0, 0, index, flags, change)


def veth_create(name0: str, name1: str):
    peer = _nlattr(VETH_INFO_PEER,
                   _ifinfo() + _nlattr(IFLA_IFNAME,
                                       name1.encode() + b"\x00"))
    linkinfo = _nlattr(IFLA_LINKINFO,
                       _nlattr(IFLA_INFO_KIND, b"veth") +
                       _nlattr(IFLA_INFO_DATA, peer))
    payload = _ifinfo() + _nlattr(IFLA_IFNAME,
                                  name0.encode() + b"\x00") + linkinfo
    _rtnl(RTM_NEWLINK,
          NLM_F_REQUEST | NLM_F_ACK | NLM_F_CREATE | NLM_F_EXCL, payload)


def link_up(name: str):
    idx = socket.if_nametoindex(name)
    _rtnl(RTM_NEWLINK, NLM_F_REQUEST | NLM_F_ACK,
          _ifinfo(idx, IFF_UP, IFF_UP))


def link_del(name: str):
    idx = socket.if_nametoindex(name)
    _rtnl(RTM_DELLINK, NLM_F_REQUEST | NLM_F_ACK, _ifinfo(idx))


# ------------------------------------------------------------ the rings
class _Ring:
    """One SPSC ring mmap'd from the XSK socket.  producer/consumer are
    u32 cells in shared memory; x86 TSO keeps desc-then-index writes
    ordered without explicit fences."""

    def __init__(self, sock_fd: int, pgoff: int, offs: Tuple[int, ...],
                 size: int, desc_sz: int):
        self.size, self.mask, self.desc_sz = size, size - 1, desc_sz
        self.mm = mmap.mmap(sock_fd, offs[2] + size * desc_sz, offset=pgoff)
        self._buf = (ctypes.c_char * len(self.mm)).from_buffer(self.mm)
        base = ctypes.addressof(self._buf)
        self._prod = ctypes.cast(base + offs[0],
                                 ctypes.POINTER(ctypes.c_uint32))
        self._cons = ctypes.cast(base + offs[1],
                                 ctypes.POINTER(ctypes.c_uint32))
        self.desc_off = offs[2]
        self.cached_prod = self._prod[0]
        self.cached_cons = self._cons[0]

    @property
    def producer(self) -> int:
        return self._prod[0]

    @producer.setter
    def producer(self, v: int):
        self._prod[0] = v & 0xFFFFFFFF

    @property
    def consumer(self) -> int:
        return self._cons[0]

    @consumer.setter
    def consumer(self, v: int):
        self._cons[0] = v & 0xFFFFFFFF

    def write_desc(self, slot: int, data: bytes):
        off = self.desc_off + (slot & self.mask) * self.desc_sz
        self.mm[off:off + len(data)] = data

    def read_desc(self, slot: int) -> bytes:
        off = self.desc_off + (slot & self.mask) * self.desc_sz
        return self.mm[off:off + self.desc_sz]


class XskSocket:
    """AF_XDP socket on one (interface, queue): UMEM + 4 rings + the
    redirect program, with the Pump source/sink API.

    mode: "auto" tries driver/zero-copy first then falls back to
    generic+copy; "copy" forces the portable path (veth/lo)."""

    def __init__(self, ifname: str, queue: int = 0, *,
                 frame_size: int = 2048, ring_size: int = 2048,
                 mode: str = "auto", pinned: bool = False,
                 attach_prog: bool = True):
        assert frame_size in (2048, 4096)
        assert ring_size & (ring_size - 1) == 0
        self.ifname, self.queue = ifname, queue
        self.frame_size = frame_size
        self.n_frames = ring_size * 2          # half RX pool, half TX pool
        umem_len = self.n_frames * frame_size
        self._pin_owner = None
        if pinned:
            # pinned UMEM: RX frames are DMA-able straight to the GPU
            import torch
            self._pin_owner = torch.empty(umem_len, dtype=torch.uint8,
                                          pin_memory=True)
            self.umem_addr = self._pin_owner.data_ptr()
            self.umem = (ctypes.c_ubyte * umem_len).from_address(
                self.umem_addr)
        else:
            self._mm = mmap.mmap(-1, umem_len)
            buf = (ctypes.c_char * umem_len).from_buffer(self._mm)
            self.umem_addr = ctypes.addressof(buf)
            self.umem = (ctypes.c_ubyte * umem_len).from_address(
                self.umem_addr)

        self.sock = socket.socket(44, socket.SOCK_RAW, 0)   # AF_XDP
        fd = self.sock.fileno()
        try:
            self._setup(fd, ifname, queue, frame_size, ring_size, mode,
                        attach_prog)
        except BaseException:
            # leave no half-built socket/prog/link behind
            for h in ("link_fd", "prog_fd", "map_fd"):
                v = getattr(self, h, -1)
                if v >= 0:
                    os.close(v)
            self.sock.close()
            raise

    def _setup(self, fd, ifname, queue, frame_size, ring_size, mode,
               attach_prog):
        self.sock.setsockopt(SOL_XDP, XDP_UMEM_REG, struct.pack(
            "<QQIIII", self.umem_addr, self.n_frames * frame_size,
            frame_size, 0, 0, 0))
        for opt in (XDP_UMEM_FILL_RING, XDP_UMEM_COMPLETION_RING,
                    XDP_RX_RING, XDP_TX_RING):
            self.sock.setsockopt(SOL_XDP, opt, struct.pack("<I", ring_size))
        offs = struct.unpack("<16Q", self.sock.getsockopt(
            SOL_XDP, XDP_MMAP_OFFSETS, 128))
        self.rx = _Ring(fd, _PGOFF_RX, offs[0:4], ring_size, 16)
        self.tx = _Ring(fd, _PGOFF_TX, offs[4:8], ring_size, 16)
        self.fill = _Ring(fd, _PGOFF_FILL, offs[8:12], ring_size, 8)
        self.comp = _Ring(fd, _PGOFF_COMP, offs[12:16], ring_size, 8)

        ifindex = socket.if_nametoindex(ifname)
        self.mode = None
        tries = ([(XDP_ZEROCOPY, XDP_FLAGS_DRV_MODE, "zerocopy"),
                  (XDP_COPY, XDP_FLAGS_SKB_MODE, "copy")]
                 if mode == "auto" else
                 [(XDP_COPY, XDP_FLAGS_SKB_MODE, "copy")])
        last = None
        for bind_fl, att_fl, name in tries:
            sa = struct.pack("<HHIII", 44, bind_fl, ifindex, queue, 0)
            r = _libc.bind(fd, ctypes.create_string_buffer(sa, len(sa)),
                           len(sa))
            if r == 0:
                self.mode, self._attach_flags = name, att_fl
                break
            last = ctypes.get_errno()
        if self.mode is None:
            raise OSError(last, f"AF_XDP bind failed on {ifname}: "
                                f"{os.strerror(last)}")

        # redirect program + map (after bind: XSKMAP requires bound fd)
        self.map_fd = self.prog_fd = self.link_fd = -1
        if attach_prog:
            self.map_fd = create_xskmap(max(4, queue + 1))
            xskmap_set(self.map_fd, queue, fd)
            self.prog_fd = load_redirect_prog(self.map_fd)
            self.link_fd = attach_xdp(self.prog_fd, ifindex,
                                      self._attach_flags)

        # frame pools: first half RX (given to fill ring), second half TX
        self._tx_free = list(range(ring_size, self.n_frames))
        self._fill_all(range(ring_size))
        self.stats = {"rx": 0, "tx": 0, "tx_dropped": 0}

    # ------------------------------------------------------------- fill
    def _fill_all(self, frames):
        p = self.fill.producer
        for fno in frames:
            self.fill.write_desc(p, struct.pack("<Q",
                                                fno * self.frame_size))
            p += 1
        self.fill.producer = p

    # --------------------------------------------------------------- RX
    def recv_batch(self, max_frames: int, timeout: float = 0.001
                   ) -> List[bytes]:
        out: List[bytes] = []
        r, _, _ = select.select([self.sock], [], [], timeout)
        if not r:
            return out
        prod = self.rx.producer
        cons = self.rx.consumer
        # u32 ring indices: subtract modulo 2^32 (the kernel wraps)
        n = min((prod - cons) & 0xFFFFFFFF, max_frames)
        refill = []
        for i in range(n):
            addr, ln, _opts = struct.unpack("<QII",
                                            self.rx.read_desc(cons + i))
            out.append(bytes(self.umem[addr:addr + ln]))
            refill.append(addr // self.frame_size)
        self.rx.consumer = cons + n
        self._fill_all(refill)
        self.stats["rx"] += n
        return out

    # --------------------------------------------------------------- TX
    def _reclaim(self):
        prod = self.comp.producer
        cons = self.comp.consumer
        for i in range((prod - cons) & 0xFFFFFFFF):
            addr = struct.unpack("<Q", self.comp.read_desc(cons + i))[0]
            self._tx_free.append(addr // self.frame_size)
        self.comp.consumer = prod

    def send_batch(self, frames: List[bytes]):
        self._reclaim()
        p = self.tx.producer
        # TX ring capacity: never advance producer past consumer+size
        room = self.tx.size - ((p - self.tx.consumer) & 0xFFFFFFFF)
        sent = 0
        for fr in frames:
            if sent >= room or not self._tx_free or \
                    len(fr) > self.frame_size:
                self.stats["tx_dropped"] += 1
                continue
            fno = self._tx_free.pop()
            addr = fno * self.frame_size
            self.umem[addr:addr + len(fr)] = fr
            self.tx.write_desc(p + sent,
                               struct.pack("<QII", addr, len(fr), 0))
            sent += 1
        self.tx.producer = p + sent
        if sent:
            self._tx_kick()
        self.stats["tx"] += sent
        return sent

    def _tx_kick(self):
        _libc.sendto(self.sock.fileno(), None, 0, socket.MSG_DONTWAIT,
                     None, 0)

    def send_batch_array(self, data, lens):
        """Array-native sink API (vectorized Pump)."""
        self._reclaim()
        p = self.tx.producer
        room = self.tx.size - ((p - self.tx.consumer) & 0xFFFFFFFF)
        sent = 0
        for i in range(len(lens)):
            ln = int(lens[i])
            if sent >= room or not self._tx_free or \
                    ln > self.frame_size:
                self.stats["tx_dropped"] += 1
                continue
            fno = self._tx_free.pop()
            addr = fno * self.frame_size
            self.umem[addr:addr + ln] = bytes(data[i, :ln])
            self.tx.write_desc(p + sent,
                               struct.pack("<QII", addr, ln, 0))
            sent += 1
        self.tx.producer = p + sent
        if sent:
            self._tx_kick()
        self.stats["tx"] += sent
        return sent

    def kernel_stats(self) -> dict:
        raw = self.sock.getsockopt(SOL_XDP, XDP_STATISTICS, 48)
        names = ("rx_dropped", "rx_invalid_descs", "tx_invalid_descs",
                 "rx_ring_full", "rx_fill_ring_empty_descs",
                 "tx_ring_empty_descs")
        vals = struct.unpack(f"<{len(raw) // 8}Q", raw)
        return dict(zip(names, vals))

    def close(self):
        for fd in (self.link_fd, self.prog_fd, self.map_fd):
            if fd >= 0:
                os.close(fd)
        self.link_fd = self.prog_fd = self.map_fd = -1
        self.sock.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
