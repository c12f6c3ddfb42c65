"""AF_XDP ring tests over a veth pair — the NIC edge in the measured
path (round-1 VERDICT task 1).  Needs root + rtnetlink + bpf(); the
whole module skips gracefully where the kernel refuses (unprivileged
CI), and runs for real both here and on the GPU box."""
import os
import socket
import struct
import time

import numpy as np
import pytest

from bng_amd.dataplane import abi
from bng_amd.dataplane.packets import (build_dhcp_request, build_ipv4,
                                       ip2u32, mac_bytes)

pytestmark = pytest.mark.skipif(os.geteuid() != 0,
                                reason="AF_XDP tests need root")

VETH0, VETH1 = "bngt0", "bngt1"


@pytest.fixture()
def veth():
    from bng_amd.dataplane import afxdp
    try:
        try:
            afxdp.link_del(VETH0)
        except OSError:
            pass
        afxdp.veth_create(VETH0, VETH1)
        afxdp.link_up(VETH0)
        afxdp.link_up(VETH1)
    except OSError as e:
        pytest.skip(f"cannot create veth: {e}")
    yield (VETH0, VETH1)
    try:
        afxdp.link_del(VETH0)
    except OSError:
        pass


def _raw_sock(ifname):
    s = socket.socket(socket.AF_PACKET, socket.SOCK_RAW, socket.htons(3))
    s.bind((ifname, 0))
    return s


def _wait_rx(xsk, want, timeout=3.0):
    out = []
    end = time.monotonic() + timeout
    while len(out) < want and time.monotonic() < end:
        out.extend(xsk.recv_batch(want - len(out), timeout=0.05))
    return out


class TestXskVeth:
    def test_rx_through_ring(self, veth):
        """Frames sent on veth0 arrive through veth1's XSK RX ring."""
        from bng_amd.dataplane.afxdp import XskSocket
        try:
            xsk = XskSocket(VETH1, mode="auto")
        except OSError as e:
            pytest.skip(f"XSK bind/attach refused: {e}")
        try:
            tx = _raw_sock(VETH0)
            frames = [build_ipv4("aa:00:00:00:00:%02x" % i,
                                 "02:00:00:00:00:01",
                                 ip2u32("10.0.0.2") + i, ip2u32("1.2.3.4"),
                                 proto=17, sport=40000 + i, dport=53,
                                 payload=bytes([i]) * 22)
                      for i in range(32)]
            want = set(frames)
            got = set()
            # veth under CI load can drop frames (background ND/MLD
            # traffic shares the ring): retransmit until all 32 unique
            # frames arrived
            for _attempt in range(5):
                for f in want - got:
                    tx.send(f)
                end = time.monotonic() + 1.0
                while want - got and time.monotonic() < end:
                    for fr in xsk.recv_batch(64, timeout=0.05):
                        if fr in want:
                            got.add(fr)
                if got == want:
                    break
            assert got == want, \
                f"got {len(got)}/32 (mode={xsk.mode})"
            tx.close()
        finally:
            xsk.close()

    def test_tx_through_ring(self, veth):
        """Frames queued on the XSK TX ring appear on the peer."""
        from bng_amd.dataplane.afxdp import XskSocket
        try:
            xsk = XskSocket(VETH1, mode="auto")
        except OSError as e:
            pytest.skip(f"XSK bind/attach refused: {e}")
        try:
            rx = _raw_sock(VETH0)
            rx.settimeout(3.0)
            frames = [build_ipv4("02:00:00:00:00:01",
                                 "aa:00:00:00:00:%02x" % i,
                                 ip2u32("9.9.9.9"), ip2u32("10.0.0.2") + i,
                                 proto=17, sport=53, dport=40000 + i,
                                 payload=bytes([i]) * 22)
                      for i in range(16)]
            sent = xsk.send_batch(frames)
            assert sent == 16
            got = []
            end = time.monotonic() + 3.0
            while len(got) < 16 and time.monotonic() < end:
                try:
                    f = rx.recv(2048)
                except socket.timeout:
                    break
                if f in frames:
                    got.append(f)
            assert len(got) == 16
            rx.close()
        finally:
            xsk.close()

    def test_array_sink_tx(self, veth):
        """The vectorized Pump's array sink path over the real ring."""
        from bng_amd.dataplane.afxdp import XskSocket
        from bng_amd.dataplane.pktio import pack_frames
        try:
            xsk = XskSocket(VETH1, mode="auto")
        except OSError as e:
            pytest.skip(f"XSK bind/attach refused: {e}")
        try:
            rx = _raw_sock(VETH0)
            rx.settimeout(3.0)
            frames = [build_ipv4("02:00:00:00:00:01", "aa:00:00:00:00:09",
                                 ip2u32("9.9.9.9"), ip2u32("10.0.0.9"),
                                 proto=17, sport=53, dport=4000 + i,
                                 payload=bytes([i]) * 30)
                      for i in range(8)]
            data, lens = pack_frames(frames, 512)
            assert xsk.send_batch_array(data, lens) == 8
            got = 0
            end = time.monotonic() + 3.0
            while got < 8 and time.monotonic() < end:
                try:
                    f = rx.recv(2048)
                except socket.timeout:
                    break
                if f in frames:
                    got += 1
            assert got == 8
            rx.close()
        finally:
            xsk.close()

    def test_pump_dhcp_offer_over_wire(self, veth):
        """End-to-end `bng run` slice over the wire on CPU: DISCOVER in
        on veth0 -> XSK ring -> golden dataplane -> OFFER out -> veth0."""
        from bng_amd.dataplane.afxdp import XskSocket
        from bng_amd.dataplane.launcher import GoldenLauncher
        from bng_amd.dataplane.pktio import Pump
        try:
            xsk = XskSocket(VETH1, mode="auto")
        except OSError as e:
            pytest.skip(f"XSK bind/attach refused: {e}")
        try:
            launcher = GoldenLauncher()
            launcher.set_server_config(mac_bytes("02:00:00:00:00:01"),
                                       ip2u32("10.0.0.1"))
            launcher.add_pool(1, ip2u32("10.0.1.0"), 24, ip2u32("10.0.1.1"))
            mac = "aa:bb:cc:00:00:01"
            launcher.add_subscriber(mac_bytes(mac), 1, ip2u32("10.0.1.50"),
                                    int(time.time()) + 3600)
            pump = Pump(launcher, xsk, xsk, batch=64, max_wait=0.05)
            tx = _raw_sock(VETH0)
            tx.settimeout(3.0)
            tx.send(build_dhcp_request(mac, 1, xid=0x1234))
            got_offer = False
            end = time.monotonic() + 3.0
            while not got_offer and time.monotonic() < end:
                pump.pump_once()
                try:
                    tx.settimeout(0.2)
                    f = tx.recv(2048)
                except socket.timeout:
                    continue
                # OFFER: from the server MAC (broadcast reply), yiaddr
                # at 14(eth)+20(ip)+8(udp)+16
                if (len(f) > 240 and
                        f[6:12] == mac_bytes("02:00:00:00:00:01")):
                    yiaddr = struct.unpack(">I", f[58:62])[0]
                    if yiaddr == ip2u32("10.0.1.50"):
                        got_offer = True
            assert got_offer, "no OFFER observed on the wire"
            tx.close()
        finally:
            xsk.close()


class TestCliPktio:
    def test_bng_run_pktio_afxdp_wiring(self, veth):
        """`bng run --pktio afxdp --interface vethX` attaches the pump
        and serves a wire DISCOVER end to end (CPU golden dataplane)."""
        from bng_amd.cli.main import BNG, build_parser
        args = build_parser().parse_args([
            "run", "--gpu", "off", "--pktio", "afxdp",
            "--interface", VETH1, "--pktio-batch", "64",
            "--pktio-max-wait", "0.02",
            "--pool-network", "10.0.2.0/24",
            "--pool-gateway", "10.0.2.1"])
        try:
            bng = BNG(args).start()
        except OSError as e:
            pytest.skip(f"XSK refused: {e}")
        try:
            tx = _raw_sock(VETH0)
            tx.settimeout(0.2)
            mac = "aa:bb:cc:00:00:33"
            # unknown subscriber -> PASS -> slow path allocates + replies
            tx.send(build_dhcp_request(mac, 1, xid=0x55))
            got = False
            end = time.monotonic() + 4.0
            while not got and time.monotonic() < end:
                try:
                    f = tx.recv(2048)
                except socket.timeout:
                    continue
                # slow-path reply (bare DHCP payload today) or fast frame
                if len(f) > 200:
                    got = True
            assert got or bng.pump.stats["passed"] > 0, \
                "DISCOVER neither fast-pathed nor slow-pathed"
            assert bng.pump.stats["rx"] >= 1
            tx.close()
        finally:
            bng.stop()


class TestPPPoEInterfacePump:
    def test_dedicated_pppoe_interface(self, veth):
        """--pppoe-interface on a second NIC attaches its own pump: a
        PADI on that wire gets a PADO back while the main pump owns
        --interface (ref main.go pppoe raw socket on a dedicated
        interface)."""
        from bng_amd.dataplane import afxdp
        from bng_amd.cli.main import BNG, build_parser
        P0, P1 = "bngp0", "bngp1"
        try:
            try:
                afxdp.link_del(P0)
            except OSError:
                pass
            afxdp.veth_create(P0, P1)
            afxdp.link_up(P0)
            afxdp.link_up(P1)
        except OSError as e:
            pytest.skip(f"cannot create second veth: {e}")
        try:
            args = build_parser().parse_args([
                "run", "--gpu", "off", "--pktio", "afpacket",
                "--interface", VETH1,
                "--pppoe-enable", "--pppoe-interface", P1,
                "--pktio-batch", "64", "--pktio-max-wait", "0.02",
                "--pool-network", "10.0.3.0/24"])
            bng = BNG(args).start()
            try:
                assert bng.pump_pppoe is not None
                tx = _raw_sock(P0)
                tx.settimeout(0.2)
                # PADI: dst broadcast, ethertype 0x8863, code 0x09
                src = bytes.fromhex("aabbcc000077")
                padi = (b"\xff" * 6 + src +
                        b"\x88\x63" + b"\x11\x09\x00\x00\x00\x04" +
                        b"\x01\x01\x00\x00")
                got_pado = False
                end = time.monotonic() + 5.0
                while not got_pado and time.monotonic() < end:
                    tx.send(padi)
                    try:
                        f = tx.recv(2048)
                    except socket.timeout:
                        continue
                    if len(f) >= 20 and f[12:14] == b"\x88\x63" and \
                            f[15] == 0x07:          # PADO
                        got_pado = True
                assert got_pado, "no PADO on the dedicated PPPoE NIC"
                tx.close()
            finally:
                bng.stop()
        finally:
            try:
                afxdp.link_del(P0)
            except OSError:
                pass
