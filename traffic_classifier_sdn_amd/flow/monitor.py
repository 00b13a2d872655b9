"""Native OpenFlow 1.3 telemetry monitor — the framework's replacement for
the reference's Ryu app (simple_monitor_13.py) with no Ryu dependency.

Combines the two behaviours the reference composes by inheritance:

* L2 learning switch (reference's SimpleSwitch13 parent): PACKET_IN ->
  learn src MAC, install a priority-1 flow matching (in_port, eth_src,
  eth_dst) with an OUTPUT action, or flood unknown destinations;
* 1 Hz stats poller (simple_monitor_13.py:31-47): OFPFlowStatsRequest +
  OFPPortStatsRequest to every live datapath each second;
* flow-stats reply emitter (simple_monitor_13.py:49-66): filter
  priority==1, sort by (in_port, eth_dst), print the TSV contract line:
  ``data\\t<time>\\t<dpid:x>\\t<in_port:x>\\t<eth_src>\\t<eth_dst>\\t<out_port:x>\\t<pkts>\\t<bytes>``

Run:  python -m traffic_classifier_sdn_amd.flow.monitor [--port 6653]
"""

from __future__ import annotations

import argparse
import asyncio
import struct
import sys
import time
from typing import Dict, Optional, TextIO

from . import openflow as of

HEADER_LINE = "time\tdatapath\tin-port\teth-src\teth-dst\tout-port\ttotal_packets\ttotal_bytes"


class Datapath:
    def __init__(self, writer: asyncio.StreamWriter):
        self.writer = writer
        self.id: Optional[int] = None
        self.mac_to_port: Dict[str, int] = {}
        self._xid = 0

    def next_xid(self) -> int:
        self._xid += 1
        return self._xid

    def send(self, data: bytes) -> None:
        self.writer.write(data)


class MonitorApp:
    """OpenFlow controller: learning switch + 1 Hz flow-stats telemetry."""

    def __init__(self, out: TextIO = sys.stdout, poll_interval: float = 1.0,
                 clock=None) -> None:
        self.out = out
        self.poll_interval = poll_interval
        self.datapaths: Dict[int, Datapath] = {}
        self.clock = clock or (lambda: int(time.time()))
        self._server: Optional[asyncio.AbstractServer] = None
        self._poll_task: Optional[asyncio.Task] = None

    # -- lifecycle -----------------------------------------------------
    async def start(self, host: str = "0.0.0.0", port: int = 6653) -> None:
        self._server = await asyncio.start_server(self._handle_conn, host, port)
        self._log(HEADER_LINE)
        self._poll_task = asyncio.create_task(self._poll_loop())

    async def stop(self) -> None:
        if self._poll_task:
            self._poll_task.cancel()
        if self._server:
            self._server.close()
            await self._server.wait_closed()

    def _log(self, line: str) -> None:
        self.out.write(line + "\n")
        self.out.flush()

    # -- poller (simple_monitor_13.py:31-47) ----------------------------
    async def _poll_loop(self) -> None:
        while True:
            for dp in list(self.datapaths.values()):
                try:
                    dp.send(of.flow_stats_request(dp.next_xid()))
                    dp.send(of.port_stats_request(dp.next_xid()))
                    await dp.writer.drain()
                except ConnectionError:
                    pass
            await asyncio.sleep(self.poll_interval)

    # -- per-switch connection ------------------------------------------
    async def _handle_conn(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter) -> None:
        dp = Datapath(writer)
        dp.send(of.hello(dp.next_xid()))
        dp.send(of.features_request(dp.next_xid()))
        await writer.drain()
        try:
            while True:
                hdr = await reader.readexactly(of.HEADER.size)
                version, msg_type, length, xid = of.parse_header(hdr)
                body = await reader.readexactly(length - of.HEADER.size) if length > of.HEADER.size else b""
                await self._dispatch(dp, msg_type, xid, body)
        except (asyncio.IncompleteReadError, ConnectionError):
            pass
        finally:
            if dp.id is not None:
                # DEAD_DISPATCHER unregistration (simple_monitor_13.py:26-29)
                self.datapaths.pop(dp.id, None)
            writer.close()

    async def _dispatch(self, dp: Datapath, msg_type: int, xid: int, body: bytes) -> None:
        try:
            await self._dispatch_inner(dp, msg_type, xid, body)
        except (ValueError, IndexError, KeyError, struct.error) as e:
            # a malformed message from one switch must not tear down its
            # channel (tests/test_openflow_golden.py fuzz coverage); the
            # framing (header length) already kept the stream in sync
            self._log_err(f"malformed OF message type={msg_type} xid={xid}: {e!r}")

    def _log_err(self, msg: str) -> None:
        print(msg, file=sys.stderr)

    async def _dispatch_inner(self, dp: Datapath, msg_type: int, xid: int, body: bytes) -> None:
        if msg_type == of.OFPT_HELLO:
            return
        if msg_type == of.OFPT_ECHO_REQUEST:
            dp.send(of.echo_reply(xid, body))
            await dp.writer.drain()
            return
        if msg_type == of.OFPT_FEATURES_REPLY:
            feats = of.decode_features_reply(body)
            dp.id = feats.datapath_id
            self.datapaths[dp.id] = dp  # MAIN_DISPATCHER registration
            # table-miss: send everything unknown to the controller
            miss = of.flow_mod_add(of.encode_match(), of.OFPP_CONTROLLER, priority=0, xid=dp.next_xid())
            dp.send(miss)
            await dp.writer.drain()
            return
        if msg_type == of.OFPT_PACKET_IN:
            await self._packet_in(dp, of.decode_packet_in(body))
            return
        if msg_type == of.OFPT_MULTIPART_REPLY:
            mp_type = int.from_bytes(body[:2], "big")
            if mp_type == of.OFPMP_FLOW:
                self._flow_stats_reply(dp, body)
            # port stats replies are ignored (parity with the reference,
            # which requests them but has no handler — SURVEY.md §2.1)
            return

    # -- learning switch (reference SimpleSwitch13 semantics) ------------
    async def _packet_in(self, dp: Datapath, pkt: of.PacketIn) -> None:
        if pkt.in_port is None or pkt.eth_src is None or pkt.eth_dst is None:
            return
        dp.mac_to_port[pkt.eth_src] = pkt.in_port
        out_port = dp.mac_to_port.get(pkt.eth_dst, of.OFPP_FLOOD)
        if out_port != of.OFPP_FLOOD:
            match = of.encode_match(in_port=pkt.in_port, eth_src=pkt.eth_src, eth_dst=pkt.eth_dst)
            dp.send(of.flow_mod_add(match, out_port, priority=1, xid=dp.next_xid()))
        dp.send(of.packet_out(pkt.in_port, out_port, pkt.data, dp.next_xid()))
        await dp.writer.drain()

    # -- telemetry emitter (simple_monitor_13.py:49-66) -------------------
    def _flow_stats_reply(self, dp: Datapath, body: bytes) -> None:
        _, stats = of.decode_flow_stats_reply(body)
        now = self.clock()
        rows = [s for s in stats if s.priority == 1]
        rows.sort(key=lambda s: (s.match.get("in_port", 0), s.match.get("eth_dst", "")))
        for s in rows:
            self._log(
                "data\t%s\t%x\t%x\t%s\t%s\t%x\t%d\t%d"
                % (
                    now,
                    dp.id or 0,
                    s.match.get("in_port", 0),
                    s.match.get("eth_src", ""),
                    s.match.get("eth_dst", ""),
                    s.out_port if s.out_port is not None else 0,
                    s.packet_count,
                    s.byte_count,
                )
            )


async def _amain(args) -> None:
    app = MonitorApp(poll_interval=args.interval)
    await app.start(args.host, args.port)
    await asyncio.Event().wait()  # run forever


def main() -> int:
    ap = argparse.ArgumentParser(description="native OpenFlow 1.3 stats monitor")
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=6653)
    ap.add_argument("--interval", type=float, default=1.0)
    args = ap.parse_args()
    try:
        asyncio.run(_amain(args))
    except KeyboardInterrupt:
        pass
    return 0


if __name__ == "__main__":
    sys.exit(main())
