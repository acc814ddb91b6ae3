"""Wire codec tests: roundtrip + hand-computed proto3 byte patterns."""
import pytest

from nerrf_amd.wire import codec


def make_event(**kw):
    defaults = dict(
        ts_sec=1_700_000_000,
        ts_nsec=123_456_789,
        pid=4242,
        tid=4243,
        comm="python3",
        syscall="openat",
        path="/app/uploads/doc_0001.dat",
        new_path="",
        flags=2,
        ret_val=-9,
        bytes=262144,
        inode="123456",
        mode=0o644,
        uid=1000,
        gid=1000,
        dependencies=["/lib/x86_64/libc.so.6"],
    )
    defaults.update(kw)
    return codec.Event(**defaults)


def test_roundtrip_single():
    ev = make_event()
    buf = codec.encode_event(ev)
    ev2 = codec.decode_event(buf)
    assert ev2 == ev


def test_roundtrip_batch():
    evs = [make_event(pid=i, path=f"/tmp/f{i}", ret_val=(-1) ** i * i) for i in range(25)]
    buf = codec.encode_event_batch(evs)
    back = codec.decode_event_batch(buf)
    assert back == evs


def test_zigzag_negative_retval():
    ev = make_event(ret_val=-2)
    buf = codec.encode_event(ev)
    # field 9 sint64, tag = (9<<3)|0 = 0x48, zigzag(-2) = 3
    assert bytes([0x48, 0x03]) in buf
    assert codec.decode_event(buf).ret_val == -2


def test_known_bytes_varint_fields():
    ev = codec.Event(pid=300)
    buf = codec.encode_event(ev)
    # field 2 varint: tag 0x10, 300 = 0xAC 0x02
    assert buf == bytes([0x10, 0xAC, 0x02])


def test_string_utf8():
    ev = codec.Event(path="/data/ünïcødé.dat")
    back = codec.decode_event(codec.encode_event(ev))
    assert back.path == "/data/ünïcødé.dat"


def test_timestamp_property():
    ev = make_event(ts_sec=10, ts_nsec=500_000_000)
    assert abs(ev.timestamp - 10.5) < 1e-9


def test_unknown_field_skipped():
    ev = make_event()
    buf = codec.encode_event(ev)
    # append unknown field 99 varint 7: tag = (99<<3)|0 = 792 -> varint 0x98 0x06
    buf2 = buf + bytes([0x98, 0x06, 0x07])
    assert codec.decode_event(buf2) == ev


def test_empty_event():
    assert codec.encode_event(codec.Event()) == b""
    assert codec.decode_event(b"") == codec.Event()


def test_against_google_protobuf_if_available():
    """Cross-check encoding against the installed protobuf runtime."""
    try:
        from google.protobuf import descriptor_pb2, descriptor_pool, message_factory
        from google.protobuf import timestamp_pb2  # noqa: F401
    except ImportError:
        pytest.skip("protobuf runtime not available")
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "nerrf_test_trace.proto"
    fdp.package = "nerrf.trace.test"
    fdp.syntax = "proto3"
    fdp.dependency.append("google/protobuf/timestamp.proto")
    msg = fdp.message_type.add()
    msg.name = "Event"
    fields = [
        ("ts", 1, "TYPE_MESSAGE", ".google.protobuf.Timestamp"),
        ("pid", 2, "TYPE_UINT32", None),
        ("tid", 3, "TYPE_UINT32", None),
        ("comm", 4, "TYPE_STRING", None),
        ("syscall", 5, "TYPE_STRING", None),
        ("path", 6, "TYPE_STRING", None),
        ("new_path", 7, "TYPE_STRING", None),
        ("flags", 8, "TYPE_INT32", None),  # enum-compatible wire format
        ("ret_val", 9, "TYPE_SINT64", None),
        ("bytes", 10, "TYPE_UINT64", None),
        ("inode", 11, "TYPE_STRING", None),
        ("mode", 12, "TYPE_UINT32", None),
        ("uid", 13, "TYPE_UINT64", None),
        ("gid", 14, "TYPE_UINT64", None),
    ]
    for name, num, ftype, tname in fields:
        f = msg.field.add()
        f.name = name
        f.number = num
        f.type = getattr(descriptor_pb2.FieldDescriptorProto, ftype)
        f.label = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
        if tname:
            f.type_name = tname
    dep = msg.field.add()
    dep.name = "dependencies"
    dep.number = 15
    dep.type = descriptor_pb2.FieldDescriptorProto.TYPE_STRING
    dep.label = descriptor_pb2.FieldDescriptorProto.LABEL_REPEATED

    pool = descriptor_pool.DescriptorPool()
    pool.Add(descriptor_pb2.FileDescriptorProto.FromString(timestamp_pb2.DESCRIPTOR.serialized_pb))
    pool.Add(fdp)
    desc = pool.FindMessageTypeByName("nerrf.trace.test.Event")
    EventMsg = message_factory.GetMessageClass(desc)

    ev = make_event()
    buf_ours = codec.encode_event(ev)
    parsed = EventMsg.FromString(buf_ours)
    assert parsed.pid == ev.pid
    assert parsed.path == ev.path
    assert parsed.ret_val == ev.ret_val
    assert parsed.ts.seconds == ev.ts_sec
    assert parsed.ts.nanos == ev.ts_nsec
    assert list(parsed.dependencies) == ev.dependencies
    # and their serialisation parses back through ours
    theirs = parsed.SerializeToString()
    back = codec.decode_event(theirs)
    assert back == ev


def test_codec_fuzz_random_bytes_no_silent_corruption():
    """Random garbage either raises ValueError or decodes to events whose
    re-encoding is stable (never crashes, never loops)."""
    import numpy as np

    rng = np.random.default_rng(0)
    for i in range(200):
        buf = bytes(rng.integers(0, 256, size=int(rng.integers(1, 80)), dtype=np.uint8))
        try:
            evs = codec.decode_event_batch(buf)
        except ValueError:
            continue
        # decodable garbage must survive a re-encode/decode cycle
        re_enc = codec.encode_event_batch(evs)
        assert codec.decode_event_batch(re_enc) == evs


def test_codec_large_values():
    ev = codec.Event(pid=2**32 - 1, bytes=2**63 - 1, ret_val=-(2**62), uid=2**40)
    back = codec.decode_event(codec.encode_event(ev))
    assert back == ev


def test_truncated_length_fields_raise():
    """Crafted/truncated length-delimited fields raise instead of silently
    decoding short payloads (matches the C++ decoder's behavior)."""
    evil = bytes([0x0A]) + b"\xff\xff\xff\xff\xff\xff\xff\xff\xff\x01"
    with pytest.raises(ValueError):
        codec.decode_event_batch(evil)
    inner = bytes([0x32, 0x20]) + b"hi"  # field 6 claims 32 bytes, has 2
    frame = bytes([0x0A, len(inner)]) + inner
    with pytest.raises(ValueError):
        codec.decode_event_batch(frame)


def test_event_roundtrip_property():
    """Property: arbitrary Events roundtrip byte-exactly through the
    hand-written codec, and the native columnar decoder agrees."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    text = st.text(max_size=40)
    u32 = st.integers(min_value=0, max_value=2**32 - 1)
    u64 = st.integers(min_value=0, max_value=2**63 - 1)

    @settings(max_examples=60, deadline=None)
    @given(
        sec=st.integers(min_value=0, max_value=2**40),
        nsec=st.integers(min_value=0, max_value=999_999_999),
        pid=u32, tid=u32, comm=text, syscall=text, path=text,
        new_path=text, ret_val=st.integers(min_value=-2**40, max_value=2**40),
        nbytes=u64,
    )
    def check(sec, nsec, pid, tid, comm, syscall, path, new_path, ret_val, nbytes):
        ev = codec.Event(ts_sec=sec, ts_nsec=nsec, pid=pid, tid=tid,
                         comm=comm, syscall=syscall, path=path,
                         new_path=new_path, ret_val=ret_val, bytes=nbytes)
        frame = codec.encode_event_batch([ev])
        back = codec.decode_event_batch(frame)
        assert len(back) == 1
        b = back[0]
        assert (b.ts_sec, b.ts_nsec, b.pid, b.tid) == (sec, nsec, pid, tid)
        assert (b.comm, b.syscall, b.path, b.new_path) == (comm, syscall, path, new_path)
        assert (b.ret_val, b.bytes) == (ret_val, nbytes)
        try:
            from nerrf_amd import _ingest
        except ImportError:
            return
        dec = _ingest.ColumnarDecoder()
        ts, pids, sysc, path_id, newp_id, nb, ret, _ = dec.decode([frame])
        assert len(ts) == 1
        assert int(pids[0]) == pid
        assert int(nb[0]) == nbytes
        assert int(ret[0]) == ret_val

    check()


def test_timestamp_negative_nanos_roundtrip():
    """Negative nanos (invalid in a well-formed Timestamp but representable):
    both codecs sign-extend to 64-bit on encode and truncate to int32 on
    decode, matching the google runtime (ADVICE r1 low finding)."""
    enc = codec.encode_timestamp(5, -7)
    sec, nsec = codec.decode_timestamp(enc)
    assert (sec, nsec) == (5, -7)
    # sign-extension => 10-byte varint for the nanos payload
    assert len(enc) > 6
    # negative seconds too (protobuf int64 two's complement)
    sec, nsec = codec.decode_timestamp(codec.encode_timestamp(-3, 1))
    assert (sec, nsec) == (-3, 1)
