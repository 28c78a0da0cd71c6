"""AMQP 0-9-1 wire codec (from scratch — no aio-pika/pika in the image).

Implements the subset of the protocol the framework needs to (a) speak to a
real RabbitMQ as a client (the reference's deployment story: BrokerManager
over aio-pika, /root/reference/llmq/core/broker.py:5-11,27-49) and (b) serve
AMQP from the in-tree broker so existing AMQP tooling can point at it.

Covered: connection/channel lifecycle, queue.declare/bind/purge/delete,
exchange.declare, basic.{qos,consume,cancel,publish,deliver,get,ack,reject,
nack,return}, content frames, RabbitMQ-flavoured field tables, heartbeat
frames. Frame format per the AMQP 0-9-1 spec §2.3: type(1) channel(2)
size(4) payload frame-end(0xCE).
"""

from __future__ import annotations

import struct
from typing import Any, Dict, List, Tuple

PROTOCOL_HEADER = b"AMQP\x00\x00\x09\x01"
FRAME_METHOD = 1
FRAME_HEADER = 2
FRAME_BODY = 3
FRAME_HEARTBEAT = 8
FRAME_END = 0xCE

DEFAULT_FRAME_MAX = 131072


class AMQPError(Exception):
    pass


# --------------------------------------------------------------- primitives

def _pack_shortstr(s: str) -> bytes:
    b = s.encode("utf-8")
    if len(b) > 255:
        raise AMQPError(f"shortstr too long: {len(b)}")
    return struct.pack("B", len(b)) + b


def _pack_longstr(b) -> bytes:
    if isinstance(b, str):
        b = b.encode("utf-8")
    return struct.pack(">I", len(b)) + b


def _pack_field_value(v: Any) -> bytes:
    if v is None:
        return b"V"
    if isinstance(v, bool):
        return b"t" + struct.pack("B", 1 if v else 0)
    if isinstance(v, int):
        if -(2**31) <= v < 2**31:
            return b"I" + struct.pack(">i", v)
        return b"l" + struct.pack(">q", v)
    if isinstance(v, float):
        return b"d" + struct.pack(">d", v)
    if isinstance(v, str):
        return b"S" + _pack_longstr(v)
    if isinstance(v, bytes):
        return b"S" + _pack_longstr(v)
    if isinstance(v, dict):
        return b"F" + pack_table(v)
    if isinstance(v, (list, tuple)):
        inner = b"".join(_pack_field_value(x) for x in v)
        return b"A" + struct.pack(">I", len(inner)) + inner
    raise AMQPError(f"cannot encode field value of type {type(v)}")


def pack_table(t: Dict[str, Any]) -> bytes:
    body = b"".join(_pack_shortstr(k) + _pack_field_value(v) for k, v in t.items())
    return struct.pack(">I", len(body)) + body


class _Reader:
    __slots__ = ("data", "pos")

    def __init__(self, data: bytes, pos: int = 0):
        self.data = data
        self.pos = pos

    def read(self, n: int) -> bytes:
        b = self.data[self.pos : self.pos + n]
        if len(b) != n:
            raise AMQPError("truncated frame payload")
        self.pos += n
        return b

    def octet(self) -> int:
        return self.read(1)[0]

    def short(self) -> int:
        return struct.unpack(">H", self.read(2))[0]

    def long(self) -> int:
        return struct.unpack(">I", self.read(4))[0]

    def longlong(self) -> int:
        return struct.unpack(">Q", self.read(8))[0]

    def shortstr(self) -> str:
        n = self.octet()
        return self.read(n).decode("utf-8", "replace")

    def longstr(self) -> bytes:
        n = self.long()
        return self.read(n)

    def field_value(self) -> Any:
        t = self.read(1)
        if t == b"t":
            return bool(self.octet())
        if t == b"b":
            return struct.unpack(">b", self.read(1))[0]
        if t == b"B":
            return self.octet()
        if t == b"s":
            return struct.unpack(">h", self.read(2))[0]
        if t == b"u":
            return self.short()
        if t == b"I":
            return struct.unpack(">i", self.read(4))[0]
        if t == b"i":
            return self.long()
        if t == b"l":
            return struct.unpack(">q", self.read(8))[0]
        if t == b"S":
            return self.longstr().decode("utf-8", "replace")
        if t == b"x":
            return self.longstr()
        if t == b"f":
            return struct.unpack(">f", self.read(4))[0]
        if t == b"d":
            return struct.unpack(">d", self.read(8))[0]
        if t == b"D":  # decimal: scale octet + long
            scale = self.octet()
            return struct.unpack(">i", self.read(4))[0] / (10 ** scale)
        if t == b"T":
            return self.longlong()
        if t == b"F":
            return self.table()
        if t == b"A":
            end = self.long() + self.pos
            out = []
            while self.pos < end:
                out.append(self.field_value())
            return out
        if t == b"V":
            return None
        raise AMQPError(f"unknown field type {t!r}")

    def table(self) -> Dict[str, Any]:
        end = self.long() + self.pos
        out: Dict[str, Any] = {}
        while self.pos < end:
            k = self.shortstr()
            out[k] = self.field_value()
        return out


# -------------------------------------------------------------- method spec
# arg types: O octet, S short, L long, Q longlong, s shortstr, l longstr,
#            T table, b bit (consecutive bits share octets)

METHODS: Dict[Tuple[int, int], Tuple[str, List[Tuple[str, str]]]] = {
    (10, 10): ("connection.start", [("version_major", "O"), ("version_minor", "O"),
                                    ("server_properties", "T"), ("mechanisms", "l"),
                                    ("locales", "l")]),
    (10, 11): ("connection.start-ok", [("client_properties", "T"), ("mechanism", "s"),
                                       ("response", "l"), ("locale", "s")]),
    (10, 30): ("connection.tune", [("channel_max", "S"), ("frame_max", "L"),
                                   ("heartbeat", "S")]),
    (10, 31): ("connection.tune-ok", [("channel_max", "S"), ("frame_max", "L"),
                                      ("heartbeat", "S")]),
    (10, 40): ("connection.open", [("virtual_host", "s"), ("reserved1", "s"),
                                   ("reserved2", "b")]),
    (10, 41): ("connection.open-ok", [("reserved1", "s")]),
    (10, 50): ("connection.close", [("reply_code", "S"), ("reply_text", "s"),
                                    ("class_id", "S"), ("method_id", "S")]),
    (10, 51): ("connection.close-ok", []),
    (20, 10): ("channel.open", [("reserved1", "s")]),
    (20, 11): ("channel.open-ok", [("reserved1", "l")]),
    (20, 40): ("channel.close", [("reply_code", "S"), ("reply_text", "s"),
                                 ("class_id", "S"), ("method_id", "S")]),
    (20, 41): ("channel.close-ok", []),
    (40, 10): ("exchange.declare", [("reserved1", "S"), ("exchange", "s"),
                                    ("type", "s"), ("passive", "b"), ("durable", "b"),
                                    ("auto_delete", "b"), ("internal", "b"),
                                    ("nowait", "b"), ("arguments", "T")]),
    (40, 11): ("exchange.declare-ok", []),
    (50, 10): ("queue.declare", [("reserved1", "S"), ("queue", "s"), ("passive", "b"),
                                 ("durable", "b"), ("exclusive", "b"),
                                 ("auto_delete", "b"), ("nowait", "b"),
                                 ("arguments", "T")]),
    (50, 11): ("queue.declare-ok", [("queue", "s"), ("message_count", "L"),
                                    ("consumer_count", "L")]),
    (50, 20): ("queue.bind", [("reserved1", "S"), ("queue", "s"), ("exchange", "s"),
                              ("routing_key", "s"), ("nowait", "b"),
                              ("arguments", "T")]),
    (50, 21): ("queue.bind-ok", []),
    (50, 30): ("queue.purge", [("reserved1", "S"), ("queue", "s"), ("nowait", "b")]),
    (50, 31): ("queue.purge-ok", [("message_count", "L")]),
    (50, 40): ("queue.delete", [("reserved1", "S"), ("queue", "s"), ("if_unused", "b"),
                                ("if_empty", "b"), ("nowait", "b")]),
    (50, 41): ("queue.delete-ok", [("message_count", "L")]),
    (60, 10): ("basic.qos", [("prefetch_size", "L"), ("prefetch_count", "S"),
                             ("global", "b")]),
    (60, 11): ("basic.qos-ok", []),
    (60, 20): ("basic.consume", [("reserved1", "S"), ("queue", "s"),
                                 ("consumer_tag", "s"), ("no_local", "b"),
                                 ("no_ack", "b"), ("exclusive", "b"), ("nowait", "b"),
                                 ("arguments", "T")]),
    (60, 21): ("basic.consume-ok", [("consumer_tag", "s")]),
    (60, 30): ("basic.cancel", [("consumer_tag", "s"), ("nowait", "b")]),
    (60, 31): ("basic.cancel-ok", [("consumer_tag", "s")]),
    (60, 40): ("basic.publish", [("reserved1", "S"), ("exchange", "s"),
                                 ("routing_key", "s"), ("mandatory", "b"),
                                 ("immediate", "b")]),
    (60, 50): ("basic.return", [("reply_code", "S"), ("reply_text", "s"),
                                ("exchange", "s"), ("routing_key", "s")]),
    (60, 60): ("basic.deliver", [("consumer_tag", "s"), ("delivery_tag", "Q"),
                                 ("redelivered", "b"), ("exchange", "s"),
                                 ("routing_key", "s")]),
    (60, 70): ("basic.get", [("reserved1", "S"), ("queue", "s"), ("no_ack", "b")]),
    (60, 71): ("basic.get-ok", [("delivery_tag", "Q"), ("redelivered", "b"),
                                ("exchange", "s"), ("routing_key", "s"),
                                ("message_count", "L")]),
    (60, 72): ("basic.get-empty", [("reserved1", "s")]),
    (60, 80): ("basic.ack", [("delivery_tag", "Q"), ("multiple", "b")]),
    (60, 90): ("basic.reject", [("delivery_tag", "Q"), ("requeue", "b")]),
    (60, 120): ("basic.nack", [("delivery_tag", "Q"), ("multiple", "b"),
                               ("requeue", "b")]),
}

NAME_TO_ID = {name: ids for ids, (name, _) in METHODS.items()}


def encode_method(name: str, **args: Any) -> bytes:
    cls, meth = NAME_TO_ID[name]
    spec = METHODS[(cls, meth)][1]
    out = [struct.pack(">HH", cls, meth)]
    bits: List[int] = []

    def flush_bits() -> None:
        while bits:
            byte = 0
            for i, b in enumerate(bits[:8]):
                byte |= (1 if b else 0) << i
            out.append(struct.pack("B", byte))
            del bits[:8]

    for argname, t in spec:
        v = args.get(argname)
        if t == "b":
            bits.append(bool(v))
            continue
        flush_bits()
        if t == "O":
            out.append(struct.pack("B", int(v or 0)))
        elif t == "S":
            out.append(struct.pack(">H", int(v or 0)))
        elif t == "L":
            out.append(struct.pack(">I", int(v or 0)))
        elif t == "Q":
            out.append(struct.pack(">Q", int(v or 0)))
        elif t == "s":
            out.append(_pack_shortstr(v or ""))
        elif t == "l":
            out.append(_pack_longstr(v if v is not None else b""))
        elif t == "T":
            out.append(pack_table(v or {}))
        else:
            raise AMQPError(f"bad spec type {t}")
    flush_bits()
    return b"".join(out)


def decode_method(payload: bytes) -> Tuple[str, Dict[str, Any]]:
    r = _Reader(payload)
    cls, meth = r.short(), r.short()
    entry = METHODS.get((cls, meth))
    if entry is None:
        raise AMQPError(f"unsupported method class={cls} method={meth}")
    name, spec = entry
    args: Dict[str, Any] = {}
    bit_byte = 0
    bit_idx = 8  # force fetch on first bit
    for argname, t in spec:
        if t == "b":
            if bit_idx >= 8:
                bit_byte = r.octet()
                bit_idx = 0
            args[argname] = bool((bit_byte >> bit_idx) & 1)
            bit_idx += 1
            continue
        bit_idx = 8
        if t == "O":
            args[argname] = r.octet()
        elif t == "S":
            args[argname] = r.short()
        elif t == "L":
            args[argname] = r.long()
        elif t == "Q":
            args[argname] = r.longlong()
        elif t == "s":
            args[argname] = r.shortstr()
        elif t == "l":
            args[argname] = r.longstr()
        elif t == "T":
            args[argname] = r.table()
    return name, args


# ------------------------------------------------------- content properties

_PROPS: List[Tuple[str, str, int]] = [  # (name, type, flag bit)
    ("content_type", "s", 15), ("content_encoding", "s", 14), ("headers", "T", 13),
    ("delivery_mode", "O", 12), ("priority", "O", 11), ("correlation_id", "s", 10),
    ("reply_to", "s", 9), ("expiration", "s", 8), ("message_id", "s", 7),
    ("timestamp", "Q", 6), ("type", "s", 5), ("user_id", "s", 4), ("app_id", "s", 3),
]


def encode_content_header(body_size: int, props: Dict[str, Any]) -> bytes:
    flags = 0
    parts: List[bytes] = []
    for name, t, bit in _PROPS:
        v = props.get(name)
        if v is None:
            continue
        flags |= 1 << bit
        if t == "s":
            parts.append(_pack_shortstr(str(v)))
        elif t == "O":
            parts.append(struct.pack("B", int(v)))
        elif t == "Q":
            parts.append(struct.pack(">Q", int(v)))
        elif t == "T":
            parts.append(pack_table(v))
    return struct.pack(">HHQH", 60, 0, body_size, flags) + b"".join(parts)


def decode_content_header(payload: bytes) -> Tuple[int, Dict[str, Any]]:
    r = _Reader(payload)
    cls = r.short()
    r.short()  # weight
    body_size = r.longlong()
    flags = r.short()
    if cls != 60:
        raise AMQPError(f"unexpected content class {cls}")
    props: Dict[str, Any] = {}
    for name, t, bit in _PROPS:
        if not flags & (1 << bit):
            continue
        if t == "s":
            props[name] = r.shortstr()
        elif t == "O":
            props[name] = r.octet()
        elif t == "Q":
            props[name] = r.longlong()
        elif t == "T":
            props[name] = r.table()
    return body_size, props


# ------------------------------------------------------------------ frames

def frame(ftype: int, channel: int, payload: bytes) -> bytes:
    return struct.pack(">BHI", ftype, channel, len(payload)) + payload + bytes([FRAME_END])


def method_frame(channel: int, name: str, **args: Any) -> bytes:
    return frame(FRAME_METHOD, channel, encode_method(name, **args))


def content_frames(
    channel: int, body: bytes, props: Dict[str, Any], frame_max: int = DEFAULT_FRAME_MAX
) -> bytes:
    out = [frame(FRAME_HEADER, channel, encode_content_header(len(body), props))]
    chunk = max(16, frame_max - 8)
    for i in range(0, len(body), chunk):
        out.append(frame(FRAME_BODY, channel, body[i : i + chunk]))
    if not body:
        pass  # zero-length body: header frame alone carries size 0
    return b"".join(out)


def heartbeat_frame() -> bytes:
    return frame(FRAME_HEARTBEAT, 0, b"")


MAX_BODY_SIZE = 64 * 1024 * 1024  # total message body cap (matches the
                                  # JSON protocol's MAX_FRAME)


async def read_frame(reader) -> Tuple[int, int, bytes]:
    """Read one frame from an asyncio StreamReader."""
    hdr = await reader.readexactly(7)
    ftype, channel, size = struct.unpack(">BHI", hdr)
    if size > 64 * 1024 * 1024:
        raise AMQPError(f"frame too large: {size}")
    payload = await reader.readexactly(size)
    end = await reader.readexactly(1)
    if end[0] != FRAME_END:
        raise AMQPError(f"bad frame end 0x{end[0]:02x}")
    return ftype, channel, payload
