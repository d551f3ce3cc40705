"""hipGraph capture helpers.

Replaces the reference's CUDA-graph tricks (`utils/cuda_graphs.py:5-76`,
per-op graphed callables) with whole-step graphs: the fused decode kernels
read their position from device memory (ops.fused_decode.DecodeContext), so a
single captured graph replays for every token of a session.
"""

from __future__ import annotations

from typing import Callable, Sequence

import torch


class GraphedCallable:
    """Capture `fn(*static_inputs)` into a hipGraph; `__call__` copies new
    inputs into the static buffers, replays, and returns the static outputs
    (callers must consume/copy them before the next replay)."""

    def __init__(self, fn: Callable, static_inputs: Sequence[torch.Tensor], warmups: int = 2):
        self.fn = fn
        self.static_inputs = list(static_inputs)
        stream = torch.cuda.Stream()
        stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(stream):
            for _ in range(warmups):
                out = fn(*self.static_inputs)
        torch.cuda.current_stream().wait_stream(stream)

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            out = fn(*self.static_inputs)
        self.static_outputs = out if isinstance(out, (tuple, list)) else (out,)
        self._single = not isinstance(out, (tuple, list))

    def __call__(self, *inputs: torch.Tensor):
        assert len(inputs) == len(self.static_inputs)
        for buf, t in zip(self.static_inputs, inputs):
            if t is not buf:
                buf.copy_(t, non_blocking=True)
        self.graph.replay()
        return self.static_outputs[0] if self._single else self.static_outputs

    def replay(self):
        self.graph.replay()
        return self.static_outputs[0] if self._single else self.static_outputs
