"""Batch iterator — array-sliced batching with async H2D double-buffering.

Replaces the reference's ``DataLoader(num_workers=4)`` (reference
main.py:162,180): our epoch data is already dense int32 arrays (builder.py),
so batching is tensor slicing — no worker processes, no per-sample collate.
On GPU, batches are staged through pinned memory and copied on a dedicated
HIP stream one batch ahead of compute.
"""

from __future__ import annotations

from typing import Iterator, Optional

import numpy as np
import torch

from ..data.builder import EpochData


class BatchIterator:
    """Iterates dict batches {'id','starts','paths','ends','label'}."""

    def __init__(
        self,
        data: EpochData,
        batch_size: int,
        shuffle: bool = True,
        seed: int = 0,
        device: Optional[torch.device] = None,
        prefetch: bool = True,
    ) -> None:
        self.data = data
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.seed = seed
        self.device = device or torch.device("cpu")
        self.prefetch = prefetch and self.device.type == "cuda"
        self._epoch = 0

        self.starts = torch.from_numpy(data.starts)
        self.paths = torch.from_numpy(data.paths)
        self.ends = torch.from_numpy(data.ends)
        self.labels = torch.from_numpy(data.labels)
        self.ids = torch.as_tensor(
            [i if i is not None else -1 for i in data.ids], dtype=torch.int64
        )
        if self.prefetch:
            self._copy_stream = torch.cuda.Stream(self.device)
            # Double-buffered PINNED staging: a fancy-indexed gather
            # (self.starts[idx]) materializes a *pageable* tensor, which
            # silently degrades the non_blocking H2D copy to a staged
            # synchronous one.  Instead each batch is gathered into one of
            # two persistent pinned buffers and copied from there; a
            # per-slot event keeps the host from overwriting a slot whose
            # copy is still in flight.
            C = self.starts.shape[1]
            self._stage = [
                {
                    "starts": torch.empty(batch_size, C, dtype=torch.int32,
                                          pin_memory=True),
                    "paths": torch.empty(batch_size, C, dtype=torch.int32,
                                         pin_memory=True),
                    "ends": torch.empty(batch_size, C, dtype=torch.int32,
                                        pin_memory=True),
                    "label": torch.empty(batch_size, dtype=torch.int64,
                                         pin_memory=True),
                }
                for _ in range(2)
            ]
            self._stage_ev = [torch.cuda.Event(), torch.cuda.Event()]
            self._stage_used = [False, False]

    def __len__(self) -> int:
        n = self.starts.shape[0]
        return (n + self.batch_size - 1) // self.batch_size

    def _order(self) -> np.ndarray:
        n = self.starts.shape[0]
        if not self.shuffle:
            return np.arange(n)
        rng = np.random.default_rng([self.seed, self._epoch, 0xBA7C])
        return rng.permutation(n)

    def __iter__(self) -> Iterator[dict]:
        order = torch.from_numpy(self._order().astype(np.int64))
        self._epoch += 1
        n = self.starts.shape[0]
        bs = self.batch_size

        def host_batch(lo: int, hi: int) -> dict:
            idx = order[lo:hi]
            return {
                "id": self.ids[idx],
                "starts": self.starts[idx],
                "paths": self.paths[idx],
                "ends": self.ends[idx],
                "label": self.labels[idx],
            }

        if not self.prefetch:
            for lo in range(0, n, bs):
                yield host_batch(lo, min(lo + bs, n))
            return

        # double-buffered async H2D on a copy stream, staged through the
        # persistent pinned buffers (a pinned source is what makes the
        # non_blocking copy actually asynchronous)
        def stage_and_copy(slot: int, lo: int, hi: int):
            if self._stage_used[slot]:
                self._stage_ev[slot].synchronize()  # copy out of this slot done
            self._stage_used[slot] = True
            idx = order[lo:hi]
            k = hi - lo
            st = self._stage[slot]
            torch.index_select(self.starts, 0, idx, out=st["starts"][:k])
            torch.index_select(self.paths, 0, idx, out=st["paths"][:k])
            torch.index_select(self.ends, 0, idx, out=st["ends"][:k])
            torch.index_select(self.labels, 0, idx, out=st["label"][:k])
            ev = self._stage_ev[slot]
            with torch.cuda.stream(self._copy_stream):
                db = {
                    "id": self.ids[idx],
                    "starts": st["starts"][:k].to(self.device, non_blocking=True),
                    "paths": st["paths"][:k].to(self.device, non_blocking=True),
                    "ends": st["ends"][:k].to(self.device, non_blocking=True),
                    "label": st["label"][:k].to(self.device, non_blocking=True),
                }
                ev.record(self._copy_stream)
            return db, ev

        pending = None
        for i, lo in enumerate(range(0, n, bs)):
            nxt = stage_and_copy(i % 2, lo, min(lo + bs, n))
            if pending is not None:
                db, ev = pending
                torch.cuda.current_stream(self.device).wait_event(ev)
                yield db
            pending = nxt
        if pending is not None:
            db, ev = pending
            torch.cuda.current_stream(self.device).wait_event(ev)
            yield db
