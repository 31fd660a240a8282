"""hipGraph-captured inference for batched code-vector export
(BASELINE.json config 5).

The forward at a fixed batch shape is captured once into a HIP graph
(torch.cuda.CUDAGraph == hipGraph on ROCm); each batch then replays the
whole kernel chain (gather+concat -> combiner MFMA -> fused attention ->
output head) with one launch, removing per-kernel launch overhead from the
launch-bound export loop.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch


class GraphedInference:
    """Capture model.forward at a fixed (B, C) shape; replay per batch.

    Tail batches smaller than B are padded with pad rows (starts=0 =>
    uniform attention over an all-pad bag) and sliced after replay.
    """

    def __init__(self, model, batch_size: int, device,
                 warmup_iters: int = 3) -> None:
        self.model = model
        self.B = batch_size
        self.C = model.option.max_path_length
        self.device = device
        model.eval()

        self.starts = torch.zeros(self.B, self.C, dtype=torch.int32, device=device)
        self.paths = torch.zeros(self.B, self.C, dtype=torch.int32, device=device)
        self.ends = torch.zeros(self.B, self.C, dtype=torch.int32, device=device)
        self.label = torch.zeros(self.B, dtype=torch.int64, device=device)

        # warmup on a side stream (allocator state), then capture
        s = torch.cuda.Stream(device)
        s.wait_stream(torch.cuda.current_stream(device))
        with torch.cuda.stream(s), torch.no_grad():
            for _ in range(warmup_iters):
                self.model(self.starts, self.paths, self.ends, self.label)
        torch.cuda.current_stream(device).wait_stream(s)

        self.graph = torch.cuda.CUDAGraph()
        with torch.no_grad():
            with torch.cuda.graph(self.graph):
                out = self.model(self.starts, self.paths, self.ends, self.label)
        self.outputs, self.code_vector, self.attention = out

    @torch.no_grad()
    def run(self, starts, paths, ends, label) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        n = starts.shape[0]
        assert n <= self.B
        for dst, src, fill in (
            (self.starts, starts, 0),
            (self.paths, paths, 0),
            (self.ends, ends, 0),
        ):
            dst[:n].copy_(src.to(torch.int32), non_blocking=True)
            if n < self.B:
                dst[n:].fill_(fill)
        self.label[:n].copy_(label, non_blocking=True)
        if n < self.B:
            self.label[n:].zero_()
        self.graph.replay()
        return (
            self.outputs[:n],
            self.code_vector[:n],
            self.attention[:n],
        )


def graphed_export_forward(model, batch_size: int, device) -> Optional[GraphedInference]:
    """Build a GraphedInference if on GPU with the HIP backend; else None."""
    if device.type != "cuda" or getattr(model, "backend", "") != "hip":
        return None
    try:
        return GraphedInference(model, batch_size, device)
    except Exception:  # capture unsupported -> eager fallback
        return None
