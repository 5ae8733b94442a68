"""Multi-stream ordering helpers + lightweight race assertions.

The reference is single-stream single-threaded (no race
detection / sanitizers). This engine runs three HIP streams — compute,
copy (pinned-host prefetch), and RCCL comm (waternet_amd/engine/fast.py) —
so cross-stream ordering bugs become possible. This module provides:

  - StreamJoin: an event-based scope that asserts (in debug mode) that a
    producer stream's work was ordered before the consumer touched the
    shared tensor, via hipEvent query.
  - ordered_copy: H2D/D2D copy on a copy stream with an event the compute
    stream waits on.
  - debug mode (WATERNET_AMD_STREAM_DEBUG=1): every join records and
    synchronizes events, turning latent races into loud failures — the
    CI-style sanitizer pass (`pytest tests -m gpu` runs one such test).
"""

import os

import torch

DEBUG = os.environ.get("WATERNET_AMD_STREAM_DEBUG", "0") == "1"


class StreamJoin:
    """Order producer-stream work before consumer-stream reads.

    Usage:
        join = StreamJoin(producer_stream)
        with torch.cuda.stream(producer_stream):
            ...  # writes shared tensors
        join.mark()                      # event on producer
        join.wait(consumer_stream)       # consumer waits (async)
    """

    def __init__(self, producer: torch.cuda.Stream):
        self.producer = producer
        self.event = torch.cuda.Event()
        self._marked = False

    def mark(self):
        self.event.record(self.producer)
        self._marked = True

    def wait(self, consumer: torch.cuda.Stream = None):
        assert self._marked, "StreamJoin.wait() before mark(): missing " \
                             "producer ordering (cross-stream race)"
        consumer = consumer or torch.cuda.current_stream()
        consumer.wait_event(self.event)
        if DEBUG:
            # sanitizer mode: force completion so misordered reads fail
            # deterministically instead of racing
            self.event.synchronize()


def ordered_copy(dst: torch.Tensor, src: torch.Tensor,
                 copy_stream: torch.cuda.Stream) -> torch.cuda.Event:
    """Copy src -> dst on copy_stream; returns the completion event the
    consumer stream must wait on before reading dst."""
    copy_stream.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(copy_stream):
        dst.copy_(src, non_blocking=True)
    ev = torch.cuda.Event()
    ev.record(copy_stream)
    return ev
