"""ETSI TS 102 232-style handover PDUs (ref pkg/intercept/exporter.go
buildHI2PDU :191-258 / buildHI3PDU :260-318 — the reference uses the
same simplified framing, noting production would be full ASN.1).

Wire format (big-endian):
  common header: version(1)=0x02, handover(1)=0x02|0x03, LIID bytes,
  NUL, seq u64, timestamp-ms u64
  HI2 body: payload_len u32, JSON IRI payload
  HI3 body: dir_len u8 + direction, src_ip_len u8 + src_ip,
  src_port u16, dst_ip_len u8 + dst_ip, dst_port u16, proto u8,
  payload_len u32, payload

Sequence numbers are per-LIID so a mediation device can detect loss on
each interception independently.  Transport is a pluggable
send(bytes); decoders are provided for tests and for a mediation-side
consumer."""
from __future__ import annotations

import json
import socket
import struct
import time
from typing import Dict, List, Optional

HI2 = 0x02
HI3 = 0x03
VERSION = 0x02


class ETSIExporter:
    """Builds + delivers HI2 (IRI) and HI3 (CC) PDUs.  `send` is any
    bytes sink; `connect_tcp(host, port)` wires a real mediation link
    (ref ETSIExporter.Connect exporter.go:73-101)."""

    def __init__(self, send=None, country_code: str = "XX"):
        self.send = send or (lambda b: None)
        self.country_code = country_code
        self.sequences: Dict[str, int] = {}
        self.sent_iri = 0
        self.sent_cc = 0
        self._sock: Optional[socket.socket] = None

    def connect_tcp(self, host: str, port: int, timeout: float = 5.0):
        s = socket.create_connection((host, port), timeout=timeout)
        self._sock = s
        self.send = s.sendall
        return self

    def close(self):
        if self._sock is not None:
            try:
                self._sock.close()
            finally:
                self._sock = None

    def _next_seq(self, liid: str) -> int:
        seq = self.sequences.get(liid, 0)
        self.sequences[liid] = seq + 1
        return seq

    def _header(self, handover: int, liid: str, seq: int,
                ts: float) -> bytes:
        return (bytes([VERSION, handover]) + liid.encode() + b"\x00" +
                struct.pack(">QQ", seq, int(ts * 1000)))

    def build_hi2(self, liid: str, event_type: str, session_id: str,
                  subscriber: str = "", src_ip: str = "",
                  dst_ip: str = "", src_port: int = 0,
                  dst_port: int = 0, protocol: int = 0,
                  ts: Optional[float] = None) -> bytes:
        ts = ts if ts is not None else time.time()
        payload = json.dumps({
            "event_type": event_type, "timestamp": ts,
            "session_id": session_id, "subscriber_id": subscriber,
            "source_ip": src_ip, "dest_ip": dst_ip,
            "source_port": src_port, "dest_port": dst_port,
            "protocol": protocol,
            "country_code": self.country_code}).encode()
        return (self._header(HI2, liid, self._next_seq(liid), ts) +
                struct.pack(">I", len(payload)) + payload)

    def build_hi3(self, liid: str, direction: str, src_ip: str,
                  dst_ip: str, src_port: int, dst_port: int,
                  protocol: int, payload: bytes,
                  ts: Optional[float] = None) -> bytes:
        ts = ts if ts is not None else time.time()
        buf = self._header(HI3, liid, self._next_seq(liid), ts)
        buf += bytes([len(direction)]) + direction.encode()
        src = socket.inet_aton(src_ip) if src_ip else b""
        dst = socket.inet_aton(dst_ip) if dst_ip else b""
        buf += bytes([len(src)]) + src + struct.pack(">H", src_port)
        buf += bytes([len(dst)]) + dst + struct.pack(">H", dst_port)
        buf += bytes([protocol])
        buf += struct.pack(">I", len(payload)) + payload
        return buf

    # -------------------------------------------- Manager integration
    def export(self, rec):
        """IRI record from the intercept Manager -> HI2 PDU."""
        d = rec.details
        self.send(self.build_hi2(
            d.get("liid", rec.warrant_id), rec.record_type,
            d.get("session_id", ""), rec.subscriber,
            src_ip=d.get("src_ip", rec.ip), dst_ip=d.get("dst_ip", ""),
            src_port=int(d.get("src_port", 0) or 0),
            dst_port=int(d.get("dst_port", 0) or 0),
            protocol=int(d.get("protocol", 0) or 0),
            ts=rec.timestamp))
        self.sent_iri += 1

    def export_cc(self, rec, payload: bytes):
        """CC record from Manager.record_cc -> HI3 PDU."""
        d = rec.details
        self.send(self.build_hi3(
            d.get("liid", rec.warrant_id), d.get("direction", "up"),
            d.get("src_ip", ""), d.get("dst_ip", ""),
            int(d.get("src_port", 0) or 0),
            int(d.get("dst_port", 0) or 0),
            int(d.get("protocol", 0) or 0), payload, ts=rec.timestamp))
        self.sent_cc += 1


def decode_pdu(buf: bytes) -> dict:
    """Mediation-side decoder for both PDU kinds (test double for the
    LEMF; raises ValueError on malformed frames)."""
    if len(buf) < 2 or buf[0] != VERSION:
        raise ValueError("bad version")
    handover = buf[1]
    nul = buf.index(0, 2)
    liid = buf[2:nul].decode()
    off = nul + 1
    seq, ts_ms = struct.unpack_from(">QQ", buf, off)
    off += 16
    out = {"handover": handover, "liid": liid, "seq": seq,
           "timestamp": ts_ms / 1000.0}
    if handover == HI2:
        (plen,) = struct.unpack_from(">I", buf, off)
        off += 4
        out["iri"] = json.loads(buf[off:off + plen])
        return out
    if handover != HI3:
        raise ValueError(f"unknown handover {handover}")
    dlen = buf[off]
    off += 1
    out["direction"] = buf[off:off + dlen].decode()
    off += dlen
    slen = buf[off]
    off += 1
    out["src_ip"] = socket.inet_ntoa(buf[off:off + slen]) if slen else ""
    off += slen
    (out["src_port"],) = struct.unpack_from(">H", buf, off)
    off += 2
    dlen2 = buf[off]
    off += 1
    out["dst_ip"] = socket.inet_ntoa(buf[off:off + dlen2]) if dlen2 else ""
    off += dlen2
    (out["dst_port"],) = struct.unpack_from(">H", buf, off)
    off += 2
    out["protocol"] = buf[off]
    off += 1
    (plen,) = struct.unpack_from(">I", buf, off)
    off += 4
    out["payload"] = buf[off:off + plen]
    return out


def split_stream(data: bytes) -> List[bytes]:
    """Split a concatenated PDU byte stream into frames (TCP delivery
    has no record boundaries; the length fields provide them)."""
    frames = []
    off = 0
    while off + 2 <= len(data):
        nul = data.index(0, off + 2)
        body = nul + 1 + 16
        handover = data[off + 1]
        if handover == HI2:
            (plen,) = struct.unpack_from(">I", data, body)
            end = body + 4 + plen
        else:
            p = body
            p += 1 + data[p]                       # direction
            p += 1 + data[p]                       # src ip
            p += 2                                 # src port
            p += 1 + data[p]                       # dst ip
            p += 2 + 1                             # dst port + proto
            (plen,) = struct.unpack_from(">I", data, p)
            end = p + 4 + plen
        frames.append(data[off:end])
        off = end
    return frames
