"""Wire-compatible tracker simulator: serves nerrf.trace.Tracker/StreamEvents.

Stands in for the eBPF tracker daemon (reference behavior contract:
tracker/cmd/tracker/main.go — gRPC server-streaming, per-client buffered
fan-out with drop-on-slow-client) so the full serving path is testable in
this container.  Streams either a recorded trace (EventArray) or a live
synthetic scenario, at a configurable replay rate.

Uses grpcio generic handlers + the hand-written wire codec (no protoc).
"""
from __future__ import annotations

import queue
import threading
import time
from concurrent import futures
from typing import Iterator, Optional

import grpc

from ..data.trace import SYSCALL_NAMES, EventArray
from ..wire import codec


def _event_array_to_wire(arr: EventArray) -> Iterator[codec.Event]:
    for i in range(len(arr)):
        path = arr.paths.lookup(int(arr.path_id[i])) if arr.path_id[i] >= 0 else ""
        newp = arr.paths.lookup(int(arr.new_path_id[i])) if arr.new_path_id[i] >= 0 else ""
        comm = arr.comms.lookup(int(arr.comm_id[i])) if arr.comm_id[i] >= 0 else ""
        ts = float(arr.ts[i])
        yield codec.Event(
            ts_sec=int(ts),
            ts_nsec=int((ts - int(ts)) * 1e9),
            pid=int(arr.pid[i]),
            tid=int(arr.pid[i]),
            comm=comm,
            syscall=SYSCALL_NAMES.get(int(arr.syscall[i]), "unknown"),
            path=path,
            new_path=newp,
            bytes=int(arr.nbytes[i]),
            ret_val=int(arr.ret_val[i]),
        )


class TrackerSimServer:
    """gRPC server streaming EventBatch frames to every connected client."""

    def __init__(
        self,
        trace: EventArray,
        address: str = "127.0.0.1:0",
        rate_multiplier: float = 0.0,  # 0 => as fast as possible
        batch_size: int = 64,
        client_buffer: int = 100,  # reference drops on full per-client buffer
        wait_for_first_client: bool = True,
    ) -> None:
        self.trace = trace
        self.rate_multiplier = rate_multiplier
        self.batch_size = batch_size
        self.client_buffer = client_buffer
        self.wait_for_first_client = wait_for_first_client
        self._clients: list[queue.Queue] = []
        self._clients_lock = threading.Lock()
        self._stop = threading.Event()
        self.dropped_batches = 0

        self._server = grpc.server(futures.ThreadPoolExecutor(max_workers=8))
        handler = grpc.method_handlers_generic_handler(
            "nerrf.trace.Tracker",
            {
                "StreamEvents": grpc.unary_stream_rpc_method_handler(
                    self._stream_events,
                    request_deserializer=codec.decode_empty,
                    response_serializer=lambda frame: frame,  # already bytes
                )
            },
        )
        self._server.add_generic_rpc_handlers((handler,))
        self.port = self._server.add_insecure_port(address)
        self._pump_thread: Optional[threading.Thread] = None

    @property
    def address(self) -> str:
        return f"127.0.0.1:{self.port}"

    def start(self) -> None:
        self._server.start()
        self._pump_thread = threading.Thread(target=self._pump, daemon=True)
        self._pump_thread.start()

    def stop(self, grace: float = 0.5) -> None:
        self._stop.set()
        self._server.stop(grace)
        if self._pump_thread:
            self._pump_thread.join(timeout=2.0)

    def _stream_events(self, request, context) -> Iterator[bytes]:
        q: queue.Queue = queue.Queue(maxsize=self.client_buffer)
        with self._clients_lock:
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
            with self._clients_lock:
                if q in self._clients:
                    self._clients.remove(q)

    def _pump(self) -> None:
        """Broadcast the trace as EventBatch frames; non-blocking per client."""
        if self.wait_for_first_client:
            while not self._stop.is_set():
                with self._clients_lock:
                    if self._clients:
                        break
                time.sleep(0.01)
        batch = []
        t_prev = None
        for ev in _event_array_to_wire(self.trace):
            if self._stop.is_set():
                return
            if self.rate_multiplier > 0 and t_prev is not None:
                dt = (ev.timestamp - t_prev) / self.rate_multiplier
                if dt > 0:
                    time.sleep(min(dt, 0.5))
            t_prev = ev.timestamp
            batch.append(ev)
            if len(batch) >= self.batch_size:
                self._broadcast(codec.encode_event_batch(batch))
                batch = []
        if batch:
            self._broadcast(codec.encode_event_batch(batch))
        # signal end of stream
        with self._clients_lock:
            for q in self._clients:
                try:
                    q.put_nowait(None)
                except queue.Full:
                    pass

    def _broadcast(self, frame: bytes) -> None:
        with self._clients_lock:
            for q in self._clients:
                try:
                    q.put_nowait(frame)  # reference semantics: drop when slow
                except queue.Full:
                    self.dropped_batches += 1
