"""Property-based invariants (hypothesis) for the wire codecs and
allocators — the randomized complement to the example-based suites."""
import ipaddress

import struct

from hypothesis import given, settings, strategies as st

from bng_amd.dhcp import message as dm
from bng_amd.radius import packet as rp
from bng_amd.allocator.bitmap import BitmapAllocator, PoolExhaustedError
from bng_amd.dataplane.abi import fnv1a64, mix64


macs = st.binary(min_size=6, max_size=6)
u32 = st.integers(min_value=0, max_value=0xFFFFFFFF)


class TestDHCPCodecProperties:
    @given(mac=macs, msg_type=st.integers(1, 8), xid=u32,
           req_ip=u32, ciaddr=u32, giaddr=u32,
           circuit=st.binary(max_size=40),
           vendor=st.text(max_size=24))
    @settings(max_examples=200, deadline=None)
    def test_encode_decode_roundtrip(self, mac, msg_type, xid, req_ip,
                                     ciaddr, giaddr, circuit, vendor):
        m = dm.build_request(mac, msg_type, xid=xid,
                             requested_ip=req_ip, ciaddr=ciaddr,
                             giaddr=giaddr, circuit_id=circuit,
                             vendor_class=vendor)
        d = dm.DHCPMessage.decode(m.encode())
        assert d.msg_type == msg_type
        assert d.xid == xid
        assert d.chaddr[:6] == mac
        assert d.ciaddr == ciaddr and d.giaddr == giaddr
        assert d.requested_ip == req_ip
        assert d.circuit_id() == circuit
        if vendor:
            assert d.vendor_class == vendor

    @given(data=st.binary(max_size=600))
    @settings(max_examples=300, deadline=None)
    def test_decode_never_crashes(self, data):
        try:
            dm.DHCPMessage.decode(data)
        except (ValueError, IndexError, KeyError,
                __import__('struct').error):
            pass


class TestRadiusCodecProperties:
    @given(code=st.integers(1, 5), ident=st.integers(0, 255),
           attrs=st.lists(st.tuples(st.integers(1, 200),
                                    st.binary(min_size=0, max_size=120)),
                          max_size=8))
    @settings(max_examples=200, deadline=None)
    def test_attr_roundtrip(self, code, ident, attrs):
        p = rp.Packet(code, ident, rp.random_authenticator())
        for t, v in attrs:
            p.add(t, v)
        d = rp.Packet.decode(p.encode())
        assert d.code == code and d.identifier == ident
        assert [(t, bytes(v)) for t, v in d.attributes] == \
            [(t, v) for t, v in attrs]


class TestAllocatorProperties:
    @given(st.data())
    @settings(max_examples=60, deadline=None)
    def test_no_double_allocation_and_release_reuse(self, data):
        a = BitmapAllocator("10.7.0.0/28", 32)
        live = {}
        for step in range(data.draw(st.integers(1, 40))):
            op = data.draw(st.sampled_from(["alloc", "release"]))
            if op == "alloc":
                sid = f"s{data.draw(st.integers(0, 20))}"
                try:
                    ip = a.allocate(sid)
                except PoolExhaustedError:
                    assert len(live) >= 13     # /28 minus reservations
                    continue
                if sid in live:
                    assert live[sid] == ip     # idempotent
                else:
                    assert ip not in live.values()
                    live[sid] = ip
            elif live:
                sid = data.draw(st.sampled_from(sorted(live)))
                a.release(sid)
                del live[sid]
        # internal view matches the model
        for sid, ip in live.items():
            assert a.lookup(sid) == ip


class TestHashProperties:
    @given(st.binary(max_size=64))
    @settings(max_examples=200, deadline=None)
    def test_fnv_python_matches_reference_vector(self, b):
        # incremental recomputation equals one-shot (associativity of
        # the fold) and stays in u64
        h = fnv1a64(b)
        assert 0 <= h < 1 << 64
        step = 0xcbf29ce484222325
        for byte in b:
            step = ((step ^ byte) * 0x100000001b3) % (1 << 64)
        assert step == h

    @given(st.integers(0, (1 << 64) - 1), st.integers(1, 64))
    @settings(max_examples=200, deadline=None)
    def test_mix64_shard_stability(self, key, shards):
        # owner assignment is a pure function and in range
        o = mix64(key) % shards
        assert 0 <= o < shards
        assert o == mix64(key) % shards


class TestEpochAllocatorProperties:
    @given(st.data())
    @settings(max_examples=40, deadline=None)
    def test_epoch_model(self, data):
        """Stateful model: active leases survive exactly grace_period
        epochs without renewal; renewal extends; no address is ever
        shared by two live subscribers."""
        from bng_amd.allocator.epoch_bitmap import EpochBitmapAllocator
        a = EpochBitmapAllocator("10.9.0.0/27", 32, grace_period=1)
        live = {}          # sid -> (ip, epochs_since_renewal)
        for _ in range(data.draw(st.integers(1, 30))):
            op = data.draw(st.sampled_from(
                ["alloc", "renew", "advance", "release"]))
            if op == "alloc":
                sid = f"s{data.draw(st.integers(0, 12))}"
                try:
                    ip = a.allocate(sid)
                except Exception:
                    continue
                if sid not in live:
                    assert ip not in [v[0] for v in live.values()]
                live[sid] = (ip, 0)
            elif op == "renew" and live:
                sid = data.draw(st.sampled_from(sorted(live)))
                a.renew(sid)
                live[sid] = (live[sid][0], 0)
            elif op == "advance":
                a.advance_epoch()
                live = {s: (ip, n + 1) for s, (ip, n) in live.items()
                        if n + 1 <= 1}          # grace_period=1
            elif live:
                sid = data.draw(st.sampled_from(sorted(live)))
                a.release(sid)
                del live[sid]
        for sid, (ip, _) in live.items():
            assert a.lookup(sid) == ip


class TestCRDTProperties:
    @given(st.lists(st.tuples(st.integers(0, 2),      # replica
                              st.integers(0, 1),      # op: put/delete
                              st.integers(0, 5),      # key id
                              st.binary(min_size=1, max_size=8)),
                    max_size=40))
    @settings(max_examples=60, deadline=None)
    def test_three_replicas_converge_identically(self, ops):
        """LWW CRDT: any op interleaving under partition converges to
        ONE state after full pairwise anti-entropy, regardless of sync
        order (commutativity + idempotence of merge)."""
        from bng_amd.nexus.clset import CLSetStore
        reps = [CLSetStore(f"n{i}") for i in range(3)]
        for r, op, k, v in ops:
            if op == 0:
                reps[r].put(f"k{k}", v)
            else:
                reps[r].delete(f"k{k}")
        # full mesh anti-entropy, two rounds, arbitrary order
        for _ in range(2):
            for i in range(3):
                for j in range(3):
                    if i != j:
                        reps[i].merge(reps[j].snapshot())
        states = [sorted((k, bytes(v)) for k, v in r.list("").items())
                  for r in reps]
        assert states[0] == states[1] == states[2]
        # merge is idempotent: re-merging changes nothing
        before = reps[0].snapshot()
        reps[0].merge(reps[1].snapshot())
        assert reps[0].snapshot() == before


@given(st.lists(st.tuples(st.integers(0, 0xFFFFFFFF),
                          st.integers(0, 32)), max_size=16),
       st.integers(0, 0xFFFFFFFF))
@settings(max_examples=300, deadline=None)
def test_interval_fold_equals_mask_scan(prefixes, probe_ip):
    """The launcher's prefix->interval fold must answer membership
    identically to the reference's per-prefix mask test (the LPM-trie
    membership semantics the dataplane binary-searches)."""
    from bng_amd.dataplane.abi import prefixes_to_intervals
    ranges = [(net & (0xFFFFFFFF << (32 - plen)) & 0xFFFFFFFF
               if plen else 0,
               (0xFFFFFFFF << (32 - plen)) & 0xFFFFFFFF if plen else 0)
              for net, plen in prefixes]
    iv = prefixes_to_intervals(ranges)
    # intervals are sorted and disjoint
    for a, b in zip(iv, iv[1:]):
        assert a[1] < b[0]
    linear = any((probe_ip & m) == n for n, m in ranges)
    binary = any(lo <= probe_ip <= hi for lo, hi in iv)
    assert binary == linear


@given(st.lists(st.tuples(st.sampled_from(["put", "del", "restart",
                                           "compact"]),
                          st.integers(0, 7),
                          st.binary(min_size=0, max_size=6)),
                max_size=40))
@settings(max_examples=60, deadline=None)
def test_clset_wal_restart_equivalence(ops):
    """Any interleaving of puts/deletes/compactions/restarts leaves the
    persisted CLSet exactly equal to an in-memory reference model
    (crash points modeled by restart-without-close: WAL replay)."""
    import shutil
    import tempfile
    from bng_amd.nexus.clset import CLSetStore
    d = tempfile.mkdtemp()
    try:
        s = CLSetStore("n", data_dir=d)
        model = {}
        for op, k, v in ops:
            key = f"k{k}"
            if op == "put":
                s.put(key, v)
                model[key] = v
            elif op == "del":
                s.delete(key)
                model.pop(key, None)
            elif op == "compact":
                s.compact()
            else:                      # crash + restart (no close())
                s._wal.flush()
                s = CLSetStore("n", data_dir=d)
        for key in [f"k{i}" for i in range(8)]:
            assert s.get(key) == model.get(key), key
        s.close()
        s2 = CLSetStore("n", data_dir=d)
        for key in [f"k{i}" for i in range(8)]:
            assert s2.get(key) == model.get(key), key
        s2.close()
    finally:
        shutil.rmtree(d, ignore_errors=True)


@given(st.lists(st.tuples(st.integers(1, 250),        # sub index
                          st.integers(1, 0xDFFFFFFF),  # dst ip
                          st.integers(1024, 65535),    # sport
                          st.sampled_from([6, 17])),
                min_size=1, max_size=30, unique=True),
       st.booleans())
@settings(max_examples=40, deadline=None)
def test_nat_ha_export_import_roundtrip(flows, eim_on):
    """Golden-level NAT HA: establish arbitrary flows, export, import
    into a FRESH dataplane — the re-export must describe the same flow
    set, and every flow's return packet must translate identically."""
    from bng_amd.dataplane import abi as A
    from bng_amd.dataplane.launcher import GoldenLauncher
    from bng_amd.dataplane.packets import build_ipv4, ip2u32

    def mk():
        l = GoldenLauncher()
        l.set_nat_config(flags=A.NAT_FLAG_EIM if eim_on else 0)
        for i in range(1, 251):
            l.add_subscriber_nat(ip2u32("10.0.0.0") + i,
                                 ip2u32("203.0.113.1"),
                                 1024 + (i % 60) * 1024,
                                 1024 + (i % 60) * 1024 + 1023,
                                 subscriber_id=i)
        return l

    a, b = mk(), mk()
    returns = []
    for sub, dst, sport, proto in flows:
        pkt = bytearray(build_ipv4(
            "aa:00:00:00:00:01", "02:00:00:00:00:01",
            ip2u32("10.0.0.0") + sub, dst, proto=proto,
            sport=sport, dport=443, payload=b"x" * 22))
        v = a.dp.nat44_egress(pkt)
        if v != A.FWD:
            continue
        nat_port = int.from_bytes(pkt[34:36], "big")
        ret = build_ipv4("02:00:00:00:00:01", "aa:00:00:00:00:01",
                         dst, ip2u32("203.0.113.1"), proto=proto,
                         sport=443, dport=nat_port, payload=b"y" * 22)
        returns.append((ret, ip2u32("10.0.0.0") + sub, sport))
    recs = a.export_nat_sessions()
    assert b.import_nat_sessions(recs) == len(recs)
    back = b.export_nat_sessions()
    key = lambda r: (int(r["src_ip"]), int(r["dst_ip"]),
                     int(r["src_port"]), int(r["dst_port"]),
                     int(r["protocol"]), int(r["nat_ip"]),
                     int(r["nat_port"]))
    assert sorted(map(key, recs)) == sorted(map(key, back))
    for ret, want_ip, want_port in returns:
        fb = bytearray(ret)
        assert b.dp.nat44_ingress(fb) == A.FWD
        assert int.from_bytes(fb[30:34], "big") == want_ip
        assert int.from_bytes(fb[36:38], "big") == want_port


class TestEtsiPduProperties:
    """ETSI HI2/HI3 codec: build/decode roundtrip and stream framing
    hold for arbitrary field values."""

    @given(liid=st.text(alphabet=st.characters(
               whitelist_categories=("Lu", "Ll", "Nd"),
               whitelist_characters="-_"), min_size=1, max_size=32),
           direction=st.sampled_from(["up", "down"]),
           sport=st.integers(0, 65535), dport=st.integers(0, 65535),
           proto=st.integers(0, 255),
           payload=st.binary(max_size=512),
           ts=st.floats(0, 4e9))
    @settings(max_examples=150, deadline=None)
    def test_hi3_roundtrip(self, liid, direction, sport, dport, proto,
                           payload, ts):
        from bng_amd.intercept.etsi import ETSIExporter, decode_pdu
        ex = ETSIExporter()
        pdu = ex.build_hi3(liid, direction, "10.0.1.5", "9.9.9.9",
                           sport, dport, proto, payload, ts=ts)
        d = decode_pdu(pdu)
        assert d["liid"] == liid and d["direction"] == direction
        assert d["src_port"] == sport and d["dst_port"] == dport
        assert d["protocol"] == proto and d["payload"] == payload
        assert abs(d["timestamp"] - ts) < 0.002

    @given(st.lists(st.tuples(st.sampled_from(["hi2", "hi3"]),
                              st.binary(max_size=64)),
                    min_size=1, max_size=8))
    @settings(max_examples=100, deadline=None)
    def test_stream_framing(self, kinds):
        from bng_amd.intercept.etsi import ETSIExporter, split_stream
        ex = ETSIExporter()
        frames = []
        for kind, payload in kinds:
            if kind == "hi2":
                frames.append(ex.build_hi2("L1", "ev", "s", ts=1.0))
            else:
                frames.append(ex.build_hi3("L1", "up", "1.2.3.4",
                                           "5.6.7.8", 1, 2, 6,
                                           payload, ts=1.0))
        assert split_stream(b"".join(frames)) == frames


class TestAuditSeverityProperties:
    @given(st.text(max_size=40))
    @settings(max_examples=200, deadline=None)
    def test_severity_and_category_total(self, action):
        """Every action string maps to a valid severity and category."""
        from bng_amd.audit import retention as rt
        sev = rt.action_severity(action)
        assert 0 <= sev < 8
        assert rt.severity_name(sev) != "UNKNOWN"
        assert isinstance(rt.action_category(action), str)

    @given(st.integers(-5, 12))
    @settings(max_examples=50, deadline=None)
    def test_severity_name_never_raises(self, sev):
        from bng_amd.audit.retention import severity_name
        assert isinstance(severity_name(sev), str)


class TestDnsCnameProperty:
    @given(st.lists(st.text(alphabet="abcdefghijklmnopqrstuvwxyz0123456789-",
                            min_size=1, max_size=20),
                    min_size=1, max_size=5))
    @settings(max_examples=150, deadline=None)
    def test_cname_target_roundtrip(self, labels):
        from bng_amd.dns.resolver import (build_cname_response,
                                          build_query, decode_qname)
        target = ".".join(labels)
        q = build_query("www.example.com")
        resp = build_cname_response(q, target)
        _qn, off = decode_qname(resp, 12)
        off += 4 + 2
        got, _ = decode_qname(resp, off + 10)
        assert got == target


class TestParserRobustness:
    """New binary parsers never hang or crash on garbage."""

    @given(st.binary(max_size=256))
    @settings(max_examples=300, deadline=None)
    def test_netlink_attr_parser_total(self, blob):
        from bng_amd.routing.netlink import _parse_attrs
        out = _parse_attrs(blob)
        assert isinstance(out, dict)

    @given(st.binary(max_size=128))
    @settings(max_examples=300, deadline=None)
    def test_etsi_decode_rejects_garbage(self, blob):
        from bng_amd.intercept.etsi import decode_pdu
        try:
            decode_pdu(blob)
        except (ValueError, IndexError, KeyError, struct.error):
            pass

    @given(st.binary(max_size=200))
    @settings(max_examples=300, deadline=None)
    def test_vendor_tlv_parser_total(self, blob):
        from bng_amd.ztp.bootstrap import parse_vendor_options
        assert isinstance(parse_vendor_options(blob), str)

    @given(st.binary(max_size=300))
    @settings(max_examples=200, deadline=None)
    def test_dns_query_handler_survives_garbage(self, blob):
        from bng_amd.dns.resolver import Resolver
        r = Resolver(lambda q: None)
        r.add_rule("x.example", action="block")
        out = r.handle_query(blob)
        assert out is None or isinstance(out, bytes)
