"""gRPC client for nerrf.trace.Tracker/StreamEvents.

Consumes the tracker stream (real daemon or tracker_sim) into the delta edge
store / EventArray batches; uses the hand-written wire codec.
"""
from __future__ import annotations

from typing import Callable, Iterator, List, Optional

import grpc

from ..wire import codec


def stream_events(
    address: str,
    timeout_s: Optional[float] = None,
) -> Iterator[List[codec.Event]]:
    """Yield decoded EventBatch lists from a tracker at `address`."""
    channel = grpc.insecure_channel(address)
    try:
        stub = channel.unary_stream(
            codec.STREAM_EVENTS_METHOD,
            request_serializer=lambda _: codec.encode_empty(),
            response_deserializer=lambda b: b,  # raw frames; decode below
        )
        for frame in stub(None, timeout=timeout_s):
            yield codec.decode_event_batch(frame)
    finally:
        channel.close()


def pump_into_store(
    address: str,
    store,
    max_events: Optional[int] = None,
    timeout_s: Optional[float] = None,
    on_batch: Optional[Callable[[List[codec.Event]], None]] = None,
) -> int:
    """Stream events into a DeltaGraphStore; returns events consumed."""
    n = 0
    try:
        for batch in stream_events(address, timeout_s=timeout_s):
            store.append_wire_batch(batch)
            if on_batch is not None:
                on_batch(batch)
            n += len(batch)
            if max_events is not None and n >= max_events:
                break
    except grpc.RpcError as e:  # deadline exceeded ends a bounded pump cleanly
        if e.code() not in (grpc.StatusCode.DEADLINE_EXCEEDED, grpc.StatusCode.CANCELLED):
            raise
    return n
