"""Native OpenFlow 1.3 monitor tested against a scripted fake switch
(the telemetry path must test without Ryu/Mininet/root)."""

import asyncio
import io

import pytest

from traffic_classifier_sdn_amd.flow import openflow as of
from traffic_classifier_sdn_amd.flow.monitor import HEADER_LINE, MonitorApp
from traffic_classifier_sdn_amd.flow.parser import PollStreamParser


class FakeSwitch:
    """Plays the OVS side: handshake, then answers flow-stats requests."""

    def __init__(self, dpid=1, stats=None):
        self.dpid = dpid
        self.stats = stats or []
        self.flow_mods = []
        self.packet_outs = []
        self.stats_requests = 0

    async def run(self, host, port, n_replies=2):
        reader, writer = await asyncio.open_connection(host, port)
        writer.write(of.hello(1))
        await writer.drain()
        replies = 0
        while replies < n_replies:
            hdr = await reader.readexactly(of.HEADER.size)
            version, msg_type, length, xid = of.parse_header(hdr)
            assert version == of.OFP_VERSION
            body = await reader.readexactly(length - of.HEADER.size) if length > of.HEADER.size else b""
            if msg_type == of.OFPT_FEATURES_REQUEST:
                writer.write(of.features_reply(self.dpid, xid))
                await writer.drain()
            elif msg_type == of.OFPT_FLOW_MOD:
                self.flow_mods.append(body)
            elif msg_type == of.OFPT_PACKET_OUT:
                self.packet_outs.append(body)
            elif msg_type == of.OFPT_MULTIPART_REQUEST:
                mp_type = int.from_bytes(body[:2], "big")
                if mp_type == of.OFPMP_FLOW:
                    self.stats_requests += 1
                    writer.write(of.flow_stats_reply(self.stats, xid))
                    await writer.drain()
                    replies += 1
        writer.close()


def _stats():
    return [
        of.FlowStat(1, 100, 10000, {"in_port": 1, "eth_src": "00:00:00:00:00:01", "eth_dst": "00:00:00:00:00:02"}, 2),
        of.FlowStat(1, 50, 5000, {"in_port": 2, "eth_src": "00:00:00:00:00:02", "eth_dst": "00:00:00:00:00:01"}, 1),
        of.FlowStat(0, 7, 70, {"in_port": 1}, None),  # table-miss: filtered out
    ]


def _run(coro):
    return asyncio.get_event_loop_policy().new_event_loop().run_until_complete(coro)


def test_monitor_end_to_end_tsv():
    out = io.StringIO()

    async def scenario():
        app = MonitorApp(out=out, poll_interval=0.05, clock=lambda: 1600000123)
        await app.start("127.0.0.1", 0)
        port = app._server.sockets[0].getsockname()[1]
        sw = FakeSwitch(dpid=0x2A, stats=_stats())
        await asyncio.wait_for(sw.run("127.0.0.1", port, n_replies=2), timeout=10)
        await app.stop()
        return sw

    sw = _run(scenario())
    lines = out.getvalue().splitlines()
    assert lines[0] == HEADER_LINE
    data = [l for l in lines if l.startswith("data\t")]
    # 2 priority-1 flows per reply, 2 replies; table-miss filtered
    assert len(data) >= 4
    f = data[0].split("\t")
    assert f[1] == "1600000123"
    assert f[2] == "2a"          # dpid hex (reference %x format)
    assert f[3] == "1"
    assert f[4] == "00:00:00:00:00:01"
    assert f[5] == "00:00:00:00:00:02"
    assert f[6] == "2"
    assert f[7] == "100" and f[8] == "10000"
    # sorted by (in_port, eth_dst): in_port 1 row before in_port 2 row
    assert data[1].split("\t")[3] == "2"
    # the emitted lines round-trip through the framework parser
    p = PollStreamParser()
    p.feed_many(data)
    assert len(p.table) == 1  # fwd + rev resolved into one flow
    assert p.records == len(data)


def test_monitor_learning_switch():
    out = io.StringIO()

    async def scenario():
        app = MonitorApp(out=out, poll_interval=10.0)
        await app.start("127.0.0.1", 0)
        port = app._server.sockets[0].getsockname()[1]

        async def switch():
            reader, writer = await asyncio.open_connection("127.0.0.1", port)
            writer.write(of.hello(1))
            flow_mods, packet_outs = [], []
            # handshake then inject two PACKET_INs
            sent = False
            while len(packet_outs) < 2:
                hdr = await reader.readexactly(of.HEADER.size)
                _, msg_type, length, xid = of.parse_header(hdr)
                body = await reader.readexactly(length - of.HEADER.size) if length > of.HEADER.size else b""
                if msg_type == of.OFPT_FEATURES_REQUEST:
                    writer.write(of.features_reply(7, xid))
                    await writer.drain()
                elif msg_type == of.OFPT_FLOW_MOD:
                    flow_mods.append(body)
                    if not sent:
                        # table-miss installed: host A -> B (unknown B: flood)
                        writer.write(of.packet_in(1, "00:00:00:00:00:0a", "00:00:00:00:00:0b"))
                        # B -> A (A is known now: flow install + packet out)
                        writer.write(of.packet_in(2, "00:00:00:00:00:0b", "00:00:00:00:00:0a"))
                        await writer.drain()
                        sent = True
                elif msg_type == of.OFPT_PACKET_OUT:
                    packet_outs.append(body)
            writer.close()
            return flow_mods, packet_outs

        fm, po = await asyncio.wait_for(switch(), timeout=10)
        await app.stop()
        return fm, po

    flow_mods, packet_outs = _run(scenario())
    # table-miss (priority 0) + learned flow (priority 1, B->A)
    assert len(flow_mods) == 2
    assert len(packet_outs) == 2
    # learned flow matches in_port=2 src=B dst=A with output port 1
    match, _ = of.decode_match(flow_mods[1], 40)
    assert match["in_port"] == 2
    assert match["eth_src"] == "00:00:00:00:00:0b"
    assert match["eth_dst"] == "00:00:00:00:00:0a"


def test_wire_format_round_trips():
    m = of.encode_match(in_port=3, eth_src="aa:bb:cc:dd:ee:ff", eth_dst="11:22:33:44:55:66")
    assert len(m) % 8 == 0
    fields, off = of.decode_match(m, 0)
    assert fields == {"in_port": 3, "eth_dst": "11:22:33:44:55:66", "eth_src": "aa:bb:cc:dd:ee:ff"}
    assert off == len(m)

    stats = _stats()[:2]
    reply = of.flow_stats_reply(stats, xid=9)
    v, t, l, xid = of.parse_header(reply)
    assert (v, t, xid) == (of.OFP_VERSION, of.OFPT_MULTIPART_REPLY, 9)
    flags, decoded = of.decode_flow_stats_reply(reply[of.HEADER.size:])
    assert len(decoded) == 2
    assert decoded[0].packet_count == 100
    assert decoded[0].match["eth_src"] == "00:00:00:00:00:01"
    assert decoded[0].out_port == 2


def test_monitor_multi_switch_and_disconnect():
    """Two datapaths register, both get polled; a disconnect unregisters
    (DEAD_DISPATCHER parity) while the survivor keeps emitting."""
    out = io.StringIO()

    async def scenario():
        app = MonitorApp(out=out, poll_interval=0.05, clock=lambda: 1600000200)
        await app.start("127.0.0.1", 0)
        port = app._server.sockets[0].getsockname()[1]
        sw1 = FakeSwitch(dpid=0x11, stats=_stats())
        sw2 = FakeSwitch(dpid=0x22, stats=_stats())
        t1 = asyncio.create_task(sw1.run("127.0.0.1", port, n_replies=1))
        t2 = asyncio.create_task(sw2.run("127.0.0.1", port, n_replies=3))
        await asyncio.wait_for(t1, timeout=10)   # sw1 disconnects first
        await asyncio.sleep(0.1)
        n_after_disconnect = len(app.datapaths)
        await asyncio.wait_for(t2, timeout=10)
        await app.stop()
        return n_after_disconnect

    n_after = _run(scenario())
    assert n_after == 1  # sw1 unregistered, sw2 still live
    data = [l for l in out.getvalue().splitlines() if l.startswith("data\t")]
    dpids = {l.split("\t")[2] for l in data}
    assert dpids == {"11", "22"}
