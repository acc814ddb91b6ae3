"""Bridge between the native collector daemon (nerrfd) and consumers.

nerrfd streams length-prefixed nerrf.trace.EventBatch frames over TCP (the
hot path — capture, parse, batching, serialization — is C++).  This module:

  * `frames_from_daemon(addr)`: iterate raw frames from a nerrfd socket,
  * `pump_daemon_into_store(...)`: feed a DeltaGraphStore using the native
    columnar decoder (nerrf_amd._ingest) when built, the Python codec
    otherwise,
  * `GrpcBridge`: re-exposes a nerrfd stream as the gRPC
    nerrf.trace.Tracker/StreamEvents contract (wire-level compatible with
    upstream clients; this image ships no grpc++ so the gRPC hop is the one
    Python layer in the chain).
"""
from __future__ import annotations

import queue
import socket
import struct
import threading
from concurrent import futures
from typing import Iterator, Optional

from ..wire import codec


def frames_from_daemon(host: str, port: int, timeout_s: Optional[float] = None) -> Iterator[bytes]:
    sock = socket.create_connection((host, port), timeout=timeout_s)
    sock.settimeout(timeout_s)
    try:
        while True:
            hdr = b""
            while len(hdr) < 4:
                chunk = sock.recv(4 - len(hdr))
                if not chunk:
                    return
                hdr += chunk
            (length,) = struct.unpack("!I", hdr)
            payload = b""
            while len(payload) < length:
                chunk = sock.recv(length - len(payload))
                if not chunk:
                    return
                payload += chunk
            yield payload
    except (socket.timeout, ConnectionResetError, OSError):
        return
    finally:
        sock.close()


def pump_daemon_into_store(
    host: str,
    port: int,
    store,
    max_events: Optional[int] = None,
    timeout_s: Optional[float] = 10.0,
) -> int:
    """Stream nerrfd frames into a DeltaGraphStore. Returns events consumed.

    Uses the native columnar decoder when the _ingest extension is built
    (bulk numpy columns, no per-event Python objects)."""
    try:
        from nerrf_amd import _ingest  # type: ignore

        decoder = _ingest.ColumnarDecoder()
        have_native = True
    except ImportError:
        decoder = None
        have_native = False

    import numpy as np

    from ..data.trace import EventArray

    n = 0
    for frame in frames_from_daemon(host, port, timeout_s=timeout_s):
        if have_native:
            ts, pid, sysc, path_id, newp_id, nbytes, ret, comm = decoder.decode([frame])
            # the decoder's id space and the store tables are kept in
            # lockstep (this pump is the store's sole feeder), so the
            # columns drop straight into an EventArray — no per-event
            # string round trips
            for s in decoder.paths_since(len(store.paths)):
                store.paths.intern(s)
            for s in decoder.comms_since(len(store.comms)):
                store.comms.intern(s)
            arr = EventArray(
                paths=store.paths, comms=store.comms,
                ts=np.asarray(ts, dtype=np.float64),
                pid=np.asarray(pid, dtype=np.int64),
                syscall=np.asarray(sysc, dtype=np.int8),
                path_id=np.asarray(path_id, dtype=np.int64),
                new_path_id=np.asarray(newp_id, dtype=np.int64),
                nbytes=np.asarray(nbytes, dtype=np.int64),
                ret_val=np.asarray(ret, dtype=np.int64),
                comm_id=np.asarray(comm, dtype=np.int64),
            )
            store.append_array(arr)
            n += len(arr)
        else:
            events = codec.decode_event_batch(frame)
            store.append_wire_batch(events)
            n += len(events)
        if max_events is not None and n >= max_events:
            break
    return n


class GrpcBridge:
    """gRPC Tracker/StreamEvents server backed by a live nerrfd stream."""

    def __init__(self, daemon_host: str, daemon_port: int, address: str = "127.0.0.1:0",
                 wait_for_client: bool = True):
        import grpc

        self.daemon_host = daemon_host
        self.daemon_port = daemon_port
        self.wait_for_client = wait_for_client
        self._stop = threading.Event()
        self._clients: list[queue.Queue] = []
        self._lock = threading.Lock()
        self._server = grpc.server(futures.ThreadPoolExecutor(max_workers=8))
        handler = grpc.method_handlers_generic_handler(
            "nerrf.trace.Tracker",
            {
                "StreamEvents": grpc.unary_stream_rpc_method_handler(
                    self._stream,
                    request_deserializer=codec.decode_empty,
                    response_serializer=lambda frame: frame,
                )
            },
        )
        self._server.add_generic_rpc_handlers((handler,))
        self.port = self._server.add_insecure_port(address)

    @property
    def address(self) -> str:
        return f"127.0.0.1:{self.port}"

    def start(self) -> None:
        self._server.start()
        self._thread = threading.Thread(target=self._pump, daemon=True)
        self._thread.start()

    def stop(self, grace: float = 0.5) -> None:
        self._stop.set()
        self._server.stop(grace)

    def _pump(self) -> None:
        if self.wait_for_client:
            # nerrfd replays as soon as ITS first client (this pump) connects;
            # hold off until a gRPC consumer is attached so nothing is lost
            import time as _time

            while not self._stop.is_set():
                with self._lock:
                    if self._clients:
                        break
                _time.sleep(0.01)
        for frame in frames_from_daemon(self.daemon_host, self.daemon_port, timeout_s=30.0):
            if self._stop.is_set():
                return
            with self._lock:
                for q in self._clients:
                    try:
                        q.put_nowait(frame)
                    except queue.Full:
                        pass  # drop-on-slow-client
        with self._lock:
            for q in self._clients:
                try:
                    q.put_nowait(None)
                except queue.Full:
                    pass

    def _stream(self, request, context) -> Iterator[bytes]:
        q: queue.Queue = queue.Queue(maxsize=100)
        with self._lock:
            self._clients.append(q)
        try:
            while not self._stop.is_set():
                try:
                    frame = q.get(timeout=0.2)
                except queue.Empty:
                    continue
                if frame is None:
                    return
                yield frame
        finally:
            with self._lock:
                if q in self._clients:
                    self._clients.remove(q)
