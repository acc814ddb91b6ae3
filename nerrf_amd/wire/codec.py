"""Hand-written proto3 wire codec for the nerrf.trace schema.

This image ships no ``protoc``/``grpc_tools``, so instead of generated stubs we
implement the (tiny, frozen) wire contract directly.  Byte-compatible with the
upstream tracker stream (schema: proto/trace.proto; reference contract
/root/reference/proto/trace.proto:11-57).  The gRPC layer (grpcio) accepts raw
``bytes`` via custom (de)serializers, so these functions plug straight into
``grpc.unary_stream_rpc_method_handler``.

Hot-path batch decoding also exists natively (tracker/daemon); this module is
the reference implementation and is what the tests validate against.
"""
from __future__ import annotations

import io
from dataclasses import dataclass, field
from typing import Iterator, List, Tuple

# ---------------------------------------------------------------------------
# varint primitives
# ---------------------------------------------------------------------------


def _write_varint(out: io.BytesIO, value: int) -> None:
    if value < 0:  # proto3 int64/enum negatives: 10-byte two's complement
        value &= (1 << 64) - 1
    while True:
        b = value & 0x7F
        value >>= 7
        if value:
            out.write(bytes((b | 0x80,)))
        else:
            out.write(bytes((b,)))
            return


def _read_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        if pos >= len(buf):
            raise ValueError("truncated varint")
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7
        if shift >= 70:
            raise ValueError("varint too long")


def _zigzag_encode(value: int) -> int:
    return (value << 1) ^ (value >> 63)


def _zigzag_decode(value: int) -> int:
    return (value >> 1) ^ -(value & 1)


def _tag(field_number: int, wire_type: int) -> int:
    return (field_number << 3) | wire_type


_VARINT, _I64, _LEN, _I32 = 0, 1, 2, 5


def _write_tag(out: io.BytesIO, field_number: int, wire_type: int) -> None:
    _write_varint(out, _tag(field_number, wire_type))


def _write_len_delimited(out: io.BytesIO, field_number: int, payload: bytes) -> None:
    _write_tag(out, field_number, _LEN)
    _write_varint(out, len(payload))
    out.write(payload)


def _write_string(out: io.BytesIO, field_number: int, value: str) -> None:
    if value:
        _write_len_delimited(out, field_number, value.encode("utf-8"))


def _write_uint(out: io.BytesIO, field_number: int, value: int) -> None:
    if value:
        _write_tag(out, field_number, _VARINT)
        _write_varint(out, value)


def _skip_field(buf: bytes, pos: int, wire_type: int) -> int:
    if wire_type == _VARINT:
        _, pos = _read_varint(buf, pos)
    elif wire_type == _I64:
        if len(buf) - pos < 8:
            raise ValueError("truncated i64")
        pos += 8
    elif wire_type == _LEN:
        n, pos = _read_varint(buf, pos)
        if n > len(buf) - pos:
            raise ValueError("truncated skip")
        pos += n
    elif wire_type == _I32:
        if len(buf) - pos < 4:
            raise ValueError("truncated i32")
        pos += 4
    else:
        raise ValueError(f"unsupported wire type {wire_type}")
    return pos


# ---------------------------------------------------------------------------
# messages
# ---------------------------------------------------------------------------


@dataclass
class Event:
    """nerrf.trace.Event — one syscall observation."""

    ts_sec: int = 0  # Timestamp.seconds (field 1 of nested Timestamp)
    ts_nsec: int = 0  # Timestamp.nanos
    pid: int = 0
    tid: int = 0
    comm: str = ""
    syscall: str = ""
    path: str = ""
    new_path: str = ""
    flags: int = 0  # OpenFlags enum
    ret_val: int = 0  # sint64 (zigzag)
    bytes: int = 0
    inode: str = ""
    mode: int = 0
    uid: int = 0
    gid: int = 0
    dependencies: List[str] = field(default_factory=list)

    @property
    def timestamp(self) -> float:
        return self.ts_sec + self.ts_nsec * 1e-9


def encode_timestamp(sec: int, nsec: int) -> bytes:
    out = io.BytesIO()
    if sec:
        _write_tag(out, 1, _VARINT)
        _write_varint(out, sec)
    if nsec:
        _write_tag(out, 2, _VARINT)
        _write_varint(out, nsec)
    return out.getvalue()


def decode_timestamp(buf: bytes) -> Tuple[int, int]:
    sec = nsec = 0
    pos = 0
    while pos < len(buf):
        key, pos = _read_varint(buf, pos)
        fnum, wtype = key >> 3, key & 7
        if fnum == 1 and wtype == _VARINT:
            sec, pos = _read_varint(buf, pos)
            # protobuf int64: two's-complement interpretation of the varint
            if sec >= 1 << 63:
                sec -= 1 << 64
        elif fnum == 2 and wtype == _VARINT:
            nsec, pos = _read_varint(buf, pos)
            # protobuf int32: truncate to the low 32 bits, signed — matches
            # the C++ codec and the google runtime on sign-extended negatives
            nsec = ((nsec & 0xFFFFFFFF) ^ 0x80000000) - 0x80000000
        else:
            pos = _skip_field(buf, pos, wtype)
    return sec, nsec


def encode_event(ev: Event) -> bytes:
    out = io.BytesIO()
    if ev.ts_sec or ev.ts_nsec:
        _write_len_delimited(out, 1, encode_timestamp(ev.ts_sec, ev.ts_nsec))
    _write_uint(out, 2, ev.pid)
    _write_uint(out, 3, ev.tid)
    _write_string(out, 4, ev.comm)
    _write_string(out, 5, ev.syscall)
    _write_string(out, 6, ev.path)
    _write_string(out, 7, ev.new_path)
    _write_uint(out, 8, ev.flags)
    if ev.ret_val:
        _write_tag(out, 9, _VARINT)
        _write_varint(out, _zigzag_encode(ev.ret_val))
    _write_uint(out, 10, ev.bytes)
    _write_string(out, 11, ev.inode)
    _write_uint(out, 12, ev.mode)
    _write_uint(out, 13, ev.uid)
    _write_uint(out, 14, ev.gid)
    for dep in ev.dependencies:
        _write_len_delimited(out, 15, dep.encode("utf-8"))
    return out.getvalue()


def decode_event(buf: bytes) -> Event:
    ev = Event()
    pos = 0
    n = len(buf)
    while pos < n:
        key, pos = _read_varint(buf, pos)
        fnum, wtype = key >> 3, key & 7
        if wtype == _LEN:
            ln, pos = _read_varint(buf, pos)
            if ln > n - pos:
                raise ValueError("truncated length-delimited field")
            payload = buf[pos : pos + ln]
            pos += ln
            if fnum == 1:
                ev.ts_sec, ev.ts_nsec = decode_timestamp(payload)
            elif fnum == 4:
                ev.comm = payload.decode("utf-8", "replace")
            elif fnum == 5:
                ev.syscall = payload.decode("utf-8", "replace")
            elif fnum == 6:
                ev.path = payload.decode("utf-8", "replace")
            elif fnum == 7:
                ev.new_path = payload.decode("utf-8", "replace")
            elif fnum == 11:
                ev.inode = payload.decode("utf-8", "replace")
            elif fnum == 15:
                ev.dependencies.append(payload.decode("utf-8", "replace"))
            # unknown LEN fields skipped implicitly
        elif wtype == _VARINT:
            val, pos = _read_varint(buf, pos)
            if fnum == 2:
                ev.pid = val
            elif fnum == 3:
                ev.tid = val
            elif fnum == 8:
                ev.flags = val
            elif fnum == 9:
                ev.ret_val = _zigzag_decode(val)
            elif fnum == 10:
                ev.bytes = val
            elif fnum == 12:
                ev.mode = val
            elif fnum == 13:
                ev.uid = val
            elif fnum == 14:
                ev.gid = val
        else:
            pos = _skip_field(buf, pos, wtype)
    return ev


def encode_event_batch(events: List[Event]) -> bytes:
    out = io.BytesIO()
    for ev in events:
        _write_len_delimited(out, 1, encode_event(ev))
    return out.getvalue()


def decode_event_batch(buf: bytes) -> List[Event]:
    events: List[Event] = []
    pos = 0
    n = len(buf)
    while pos < n:
        key, pos = _read_varint(buf, pos)
        fnum, wtype = key >> 3, key & 7
        if fnum == 1 and wtype == _LEN:
            ln, pos = _read_varint(buf, pos)
            if ln > n - pos:
                raise ValueError("truncated event")
            events.append(decode_event(buf[pos : pos + ln]))
            pos += ln
        else:
            pos = _skip_field(buf, pos, wtype)
    return events


class Empty:
    """nerrf.trace.Empty (no fields).  A real instance (not None) because
    grpcio interprets a None deserializer result as a deserialization error."""

    def __eq__(self, other) -> bool:
        return isinstance(other, Empty)


def encode_empty(_msg: "Empty | None" = None) -> bytes:
    return b""


def decode_empty(buf: bytes) -> Empty:  # noqa: ARG001 - contract signature
    return Empty()


STREAM_EVENTS_METHOD = "/nerrf.trace.Tracker/StreamEvents"


def iter_batches_from_frames(frames: Iterator[bytes]) -> Iterator[List[Event]]:
    for frame in frames:
        yield decode_event_batch(frame)
