"""Minimal OpenFlow 1.3 wire protocol: exactly the subset the telemetry
monitor needs (reference: simple_monitor_13.py uses Ryu for this; this
framework speaks OF1.3 natively — handshake, learning-switch flow install,
flow-stats polling).

Only OF version 0x04 (1.3) is supported.  Encoders return bytes; decoders
take the message payload after the common header.
"""

from __future__ import annotations

import struct
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

OFP_VERSION = 0x04
HEADER = struct.Struct("!BBHI")  # version, type, length, xid

# message types
OFPT_HELLO = 0
OFPT_ERROR = 1
OFPT_ECHO_REQUEST = 2
OFPT_ECHO_REPLY = 3
OFPT_FEATURES_REQUEST = 5
OFPT_FEATURES_REPLY = 6
OFPT_PACKET_IN = 10
OFPT_PACKET_OUT = 13
OFPT_FLOW_MOD = 14
OFPT_MULTIPART_REQUEST = 18
OFPT_MULTIPART_REPLY = 19

# multipart types
OFPMP_FLOW = 1
OFPMP_PORT_STATS = 4

OFPP_ANY = 0xFFFFFFFF
OFPG_ANY = 0xFFFFFFFF
OFPP_FLOOD = 0xFFFFFFFB
OFPP_CONTROLLER = 0xFFFFFFFD
OFP_NO_BUFFER = 0xFFFFFFFF

OFPFC_ADD = 0

OFPIT_APPLY_ACTIONS = 4
OFPAT_OUTPUT = 0

# OXM
OXM_CLASS_OPENFLOW_BASIC = 0x8000
OXM_OF_IN_PORT = 0
OXM_OF_ETH_DST = 3
OXM_OF_ETH_SRC = 4


def header(msg_type: int, length: int, xid: int) -> bytes:
    return HEADER.pack(OFP_VERSION, msg_type, length, xid)


def message(msg_type: int, body: bytes = b"", xid: int = 0) -> bytes:
    return header(msg_type, HEADER.size + len(body), xid) + body


def parse_header(buf: bytes) -> Tuple[int, int, int, int]:
    return HEADER.unpack_from(buf)


# ----------------------------------------------------------------------
# OXM match encoding/decoding
# ----------------------------------------------------------------------


def _oxm(field_id: int, value: bytes) -> bytes:
    return struct.pack("!HBB", OXM_CLASS_OPENFLOW_BASIC, field_id << 1, len(value)) + value


def encode_match(
    in_port: Optional[int] = None,
    eth_src: Optional[str] = None,
    eth_dst: Optional[str] = None,
) -> bytes:
    """OF1.3 ofp_match (type=OXM), padded to a multiple of 8."""
    fields = b""
    if in_port is not None:
        fields += _oxm(OXM_OF_IN_PORT, struct.pack("!I", in_port))
    if eth_dst is not None:
        fields += _oxm(OXM_OF_ETH_DST, mac_to_bytes(eth_dst))
    if eth_src is not None:
        fields += _oxm(OXM_OF_ETH_SRC, mac_to_bytes(eth_src))
    length = 4 + len(fields)
    pad = (8 - length % 8) % 8
    return struct.pack("!HH", 1, length) + fields + b"\x00" * pad


def decode_match(buf: bytes, off: int) -> Tuple[Dict[str, object], int]:
    """Returns ({'in_port':int,'eth_src':str,'eth_dst':str,...}, next_offset)."""
    mtype, mlen = struct.unpack_from("!HH", buf, off)
    fields: Dict[str, object] = {}
    end = off + mlen
    p = off + 4
    while p + 4 <= end:
        cls, fh, flen = struct.unpack_from("!HBB", buf, p)
        fid = fh >> 1
        val = buf[p + 4 : p + 4 + flen]
        if cls == OXM_CLASS_OPENFLOW_BASIC:
            if fid == OXM_OF_IN_PORT:
                fields["in_port"] = struct.unpack("!I", val[:4])[0]
            elif fid == OXM_OF_ETH_SRC:
                fields["eth_src"] = bytes_to_mac(val[:6])
            elif fid == OXM_OF_ETH_DST:
                fields["eth_dst"] = bytes_to_mac(val[:6])
        p += 4 + flen
    return fields, off + mlen + ((8 - mlen % 8) % 8)


def mac_to_bytes(mac: str) -> bytes:
    return bytes(int(x, 16) for x in mac.split(":"))


def bytes_to_mac(b: bytes) -> str:
    return ":".join(f"{x:02x}" for x in b)


# ----------------------------------------------------------------------
# request encoders (controller -> switch)
# ----------------------------------------------------------------------


def hello(xid: int = 0) -> bytes:
    return message(OFPT_HELLO, b"", xid)


def echo_reply(xid: int, payload: bytes = b"") -> bytes:
    return message(OFPT_ECHO_REPLY, payload, xid)


def features_request(xid: int = 0) -> bytes:
    return message(OFPT_FEATURES_REQUEST, b"", xid)


def flow_stats_request(xid: int = 0) -> bytes:
    body = struct.pack("!HH4x", OFPMP_FLOW, 0)
    body += struct.pack("!B3xII4xQQ", 0xFF, OFPP_ANY, OFPG_ANY, 0, 0)
    body += encode_match()
    return message(OFPT_MULTIPART_REQUEST, body, xid)


def port_stats_request(xid: int = 0) -> bytes:
    body = struct.pack("!HH4x", OFPMP_PORT_STATS, 0)
    body += struct.pack("!I4x", OFPP_ANY)
    return message(OFPT_MULTIPART_REQUEST, body, xid)


def _apply_output(port: int) -> bytes:
    action = struct.pack("!HHIH6x", OFPAT_OUTPUT, 16, port, 0xFFFF)
    return struct.pack("!HH4x", OFPIT_APPLY_ACTIONS, 8 + len(action)) + action


def flow_mod_add(
    match: bytes, out_port: int, priority: int = 1, xid: int = 0, buffer_id: int = OFP_NO_BUFFER
) -> bytes:
    body = struct.pack(
        "!QQBBHHHIIIH2x",
        0, 0,                # cookie, cookie_mask
        0, OFPFC_ADD,        # table_id, command
        0, 0,                # idle, hard timeout
        priority,
        buffer_id,
        OFPP_ANY, OFPG_ANY,
        0,                   # flags
    )
    return message(OFPT_FLOW_MOD, body + match + _apply_output(out_port), xid)


def packet_out(in_port: int, out_port: int, data: bytes, xid: int = 0) -> bytes:
    action = struct.pack("!HHIH6x", OFPAT_OUTPUT, 16, out_port, 0xFFFF)
    body = struct.pack("!IIH6x", OFP_NO_BUFFER, in_port, len(action)) + action + data
    return message(OFPT_PACKET_OUT, body, xid)


# ----------------------------------------------------------------------
# reply decoders (switch -> controller)
# ----------------------------------------------------------------------


@dataclass
class FeaturesReply:
    datapath_id: int
    n_buffers: int
    n_tables: int


def decode_features_reply(body: bytes) -> FeaturesReply:
    dpid, n_buffers, n_tables = struct.unpack_from("!QIB", body)
    return FeaturesReply(dpid, n_buffers, n_tables)


@dataclass
class FlowStat:
    priority: int
    packet_count: int
    byte_count: int
    match: Dict[str, object] = field(default_factory=dict)
    out_port: Optional[int] = None
    duration_sec: int = 0


def decode_flow_stats_reply(body: bytes) -> Tuple[int, List[FlowStat]]:
    """Returns (multipart flags, [FlowStat...])."""
    mp_type, flags = struct.unpack_from("!HH", body, 0)
    assert mp_type == OFPMP_FLOW
    stats: List[FlowStat] = []
    off = 8
    while off + 2 <= len(body):
        (length,) = struct.unpack_from("!H", body, off)
        if length < 56 or off + length > len(body):
            break
        (dur_sec,) = struct.unpack_from("!I", body, off + 4)
        (priority,) = struct.unpack_from("!H", body, off + 12)
        cookie, pkts, byts = struct.unpack_from("!QQQ", body, off + 24)
        match, ioff = decode_match(body, off + 48)
        out_port = _first_output_port(body, ioff, off + length)
        stats.append(FlowStat(priority, pkts, byts, match, out_port, dur_sec))
        off += length
    return flags, stats


def _first_output_port(buf: bytes, off: int, end: int) -> Optional[int]:
    while off + 8 <= end:
        itype, ilen = struct.unpack_from("!HH", buf, off)
        if ilen < 8:
            break
        if itype == OFPIT_APPLY_ACTIONS:
            p = off + 8
            while p + 8 <= off + ilen:
                atype, alen = struct.unpack_from("!HH", buf, p)
                if alen < 8:
                    break
                if atype == OFPAT_OUTPUT:
                    (port,) = struct.unpack_from("!I", buf, p + 4)
                    return port
                p += alen
        off += ilen
    return None


@dataclass
class PacketIn:
    buffer_id: int
    in_port: Optional[int]
    eth_src: Optional[str]
    eth_dst: Optional[str]
    data: bytes


def decode_packet_in(body: bytes) -> PacketIn:
    buffer_id, total_len, reason, table_id, cookie = struct.unpack_from("!IHBBQ", body, 0)
    match, off = decode_match(body, 16)
    data = body[off + 2 :]  # 2 pad bytes before the frame
    eth_dst = bytes_to_mac(data[0:6]) if len(data) >= 12 else None
    eth_src = bytes_to_mac(data[6:12]) if len(data) >= 12 else None
    return PacketIn(buffer_id, match.get("in_port"), eth_src, eth_dst, data)


# switch-side encoders (used by the fake switch in tests)


def features_reply(datapath_id: int, xid: int) -> bytes:
    body = struct.pack("!QIBB2xII", datapath_id, 0, 1, 0, 0, 0)
    return message(OFPT_FEATURES_REPLY, body, xid)


def flow_stats_reply(stats: List[FlowStat], xid: int) -> bytes:
    body = struct.pack("!HH4x", OFPMP_FLOW, 0)
    for st in stats:
        match = encode_match(
            in_port=st.match.get("in_port"),
            eth_src=st.match.get("eth_src"),
            eth_dst=st.match.get("eth_dst"),
        )
        instr = _apply_output(st.out_port if st.out_port is not None else 1)
        entry_len = 48 + len(match) + len(instr)
        body += struct.pack(
            "!HB1xIIHHHH4xQQQ",
            entry_len, 0,
            st.duration_sec, 0,
            st.priority, 0, 0, 0,
            0, st.packet_count, st.byte_count,
        )
        body += match + instr
    return message(OFPT_MULTIPART_REPLY, body, xid)


def packet_in(in_port: int, eth_src: str, eth_dst: str, xid: int = 0) -> bytes:
    frame = mac_to_bytes(eth_dst) + mac_to_bytes(eth_src) + b"\x08\x00" + b"\x00" * 20
    match = encode_match(in_port=in_port)
    body = struct.pack("!IHBBQ", OFP_NO_BUFFER, len(frame), 0, 0, 0) + match + b"\x00\x00" + frame
    return message(OFPT_PACKET_IN, body, xid)
