"""Async, bucketing host->device dataloader.

Reference semantics (core/async_loader.py): AsyncLoader wraps any DataLoader;
batches are (optionally) padded up to the nearest length bucket and uploaded
to the device ahead of consumption. The XLA build needed bucketing to bound
recompilation; on eager MI355X bucketing is optional (kept for API parity and
for kernels that prefer a few static shapes, e.g. hipGraph capture) and the
upload is a pinned-memory + non_blocking copy issued on a dedicated "h2d"
HIP stream one batch ahead.
"""
import queue
import threading
from typing import Dict, List, Optional

import torch

from .dist.backend import get_comm_stream
from .utils.logger import logger
from .utils.utils import recursively_apply


def _uniform_buckets(max_length: int, num_buckets: int = 8) -> List[int]:
    """reference async_loader.py:14"""
    return [max_length // num_buckets * (i + 1) for i in range(num_buckets)]


def _get_closest_bucket(buckets: List[int], length: int) -> int:
    """smallest bucket >= length, else the largest (reference :20-33)."""
    for b in buckets:
        if b >= length:
            return b
    return buckets[-1]


class AsyncLoader:

    def __init__(self, loader, device, buckets: Optional[List[int]] = None,
                 max_length: Optional[int] = None,
                 num_buckets: Optional[int] = None,
                 pad_value_dict: Optional[Dict[str, int]] = None,
                 prefetch: int = 2):
        self.loader = loader
        self.device = torch.device(device)
        if buckets:
            self.buckets = sorted(buckets)
        elif max_length is not None:
            self.buckets = _uniform_buckets(max_length, num_buckets or 8)
        else:
            self.buckets = None
        self.pad_value_dict = pad_value_dict or {}
        self.prefetch = max(1, prefetch)

    def __len__(self):
        return len(self.loader)

    def _pad_batch(self, batch):
        if self.buckets is None or not isinstance(batch, dict):
            return batch
        out = {}
        for key, t in batch.items():
            if isinstance(t, torch.Tensor) and t.dim() >= 2:
                length = t.shape[-1]
                bucket = _get_closest_bucket(self.buckets, length)
                if bucket > length:
                    pad_val = self.pad_value_dict.get(key, 0)
                    pad = t.new_full(
                        (*t.shape[:-1], bucket - length), pad_val)
                    t = torch.cat([t, pad], dim=-1)
                    logger.debug("bucket pad %s: %d -> %d", key, length,
                                 bucket)
                elif bucket < length:
                    t = t[..., :bucket]
            out[key] = t
        return out

    def _to_device(self, batch):
        stream = get_comm_stream("h2d") if self.device.type == "cuda" \
            else None

        def move(t: torch.Tensor):
            if self.device.type == "cuda":
                if not t.is_pinned() and t.device.type == "cpu":
                    t = t.pin_memory()
                return t.to(self.device, non_blocking=True)
            return t.to(self.device)

        if stream is not None:
            with torch.cuda.stream(stream):
                moved = recursively_apply(move, batch)
            ev = torch.cuda.Event()
            ev.record(stream)
            return moved, ev
        return recursively_apply(move, batch), None

    def __iter__(self):
        q: "queue.Queue" = queue.Queue(maxsize=self.prefetch)
        stop = object()

        def worker():
            try:
                for batch in self.loader:
                    batch = self._pad_batch(batch)
                    q.put(self._to_device(batch))
            except Exception as e:  # surface worker errors to the consumer
                q.put(e)
            finally:
                q.put(stop)

        t = threading.Thread(target=worker, daemon=True)
        t.start()
        while True:
            item = q.get()
            if item is stop:
                break
            if isinstance(item, Exception):
                raise item
            batch, ev = item
            if ev is not None:
                cur = torch.cuda.current_stream()
                cur.wait_event(ev)
                # the device tensors were allocated on the side h2d stream;
                # mark them in-use by the consumer stream so the caching
                # allocator cannot recycle their blocks for the next batch's
                # copy while compute kernels still read them

                def _claim(t: torch.Tensor):
                    if t.is_cuda:
                        t.record_stream(cur)
                    return t

                recursively_apply(_claim, batch)
            yield batch


# reference-compatible names
CUDALoader = AsyncLoader
BucketingParallelLoader = AsyncLoader
