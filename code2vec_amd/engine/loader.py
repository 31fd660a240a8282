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
        # non_blocking copy actually asynchronous).  The gather runs in
        # NUMPY into the pinned buffers' views: torch's CPU index_select
        # is ~70x slower (7.9 vs 0.11 ms per field at the top11 shape)
        # and made the input pipeline the bottleneck.
        order_np = order.numpy()
        src_np = {
            "starts": self.starts.numpy(), "paths": self.paths.numpy(),
            "ends": self.ends.numpy(), "label": self.labels.numpy(),
        }
        stage_np = [
            {k: v.numpy() for k, v in s.items()} for s in self._stage
        ]

        def stage_and_copy(slot: int, lo: int, hi: int):
            if self._stage_used[slot]:
                self._stage_ev[slot].synchronize()  # copy out of this slot done
            self._stage_used[slot] = True
            idx = order[lo:hi]
            idx_np = order_np[lo:hi]
            k = hi - lo
            st = self._stage[slot]
            sn = stage_np[slot]
            for key in ("starts", "paths", "ends", "label"):
                np.take(src_np[key], idx_np, axis=0, out=sn[key][:k])
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

        def hand_over(db, ev):
            # the batch tensors were ALLOCATED on the copy stream; tell the
            # caching allocator they are consumed on the compute stream, or
            # their memory can be reused by a later copy while compute still
            # reads it (GPU memory fault at scale)
            cur = torch.cuda.current_stream(self.device)
            cur.wait_event(ev)
            for k, v in db.items():
                if k != "id":
                    v.record_stream(cur)
            return db

        pending = None
        for i, lo in enumerate(range(0, n, bs)):
            nxt = stage_and_copy(i % 2, lo, min(lo + bs, n))
            if pending is not None:
                yield hand_over(*pending)
            pending = nxt
        if pending is not None:
            yield hand_over(*pending)
