"""Serving engine: torch-eager prefill -> hipGraph-captured decode loop.

Capability parity with Triton-distributed python/triton_dist/models/
engine.py:37-189 (Engine.serve: prefill, backend switch, CUDA-graph decode,
sample). All decode-step state (tokens, KV offset, barrier epochs, flags)
is device-resident, so one captured graph replays for every generated token.
"""
from __future__ import annotations

from typing import Optional

import torch

from .dense import DenseLLM
from .kv_cache import KVCache


class Engine:
    def __init__(self, model: DenseLLM, batch: int, max_len: int,
                 use_graph: Optional[bool] = None):
        self.model = model
        self.batch = batch
        self.max_len = max_len
        if torch.cuda.is_available():
            from ..tune import maybe_enable_tunableop
            maybe_enable_tunableop()  # shipped library-GEMM algo picks
        cfg = model.cfg
        self.kv = model.make_cache(batch, max_len)
        if use_graph is None:
            use_graph = torch.cuda.is_available()
        self.use_graph = use_graph
        self.graph = None
        self._token_buf = None
        self._next_buf = None

    def _ensure_graph(self):
        if self.graph is not None:
            return
        dev = self.model.device
        self._token_buf = torch.zeros(self.batch, dtype=torch.int64,
                                      device=dev)
        # warmup on a side stream (captures allocate; cuBLAS/hipBLASLt init)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                saved = self.kv.offset.clone()
                nxt = self.model.decode_step(self._token_buf, self.kv)
                self.kv.offset.copy_(saved)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self._next_buf = self.model.decode_step(self._token_buf, self.kv)
            self._token_buf.copy_(self._next_buf)

    def decode_once(self, tokens: torch.Tensor) -> torch.Tensor:
        """One decode step for [B] tokens -> [B] next tokens."""
        if self.use_graph:
            self._ensure_graph()
            self._token_buf.copy_(tokens)
            self.graph.replay()
            return self._token_buf
        return self.model.decode_step(tokens, self.kv)

    def profile_decode(self, steps: int = 10,
                       trace_path: str | None = None) -> dict:
        """torch-profile `steps` decode replays and return per-kernel
        totals for ONE step ({name: {us_per_step, calls}}); optionally
        export a chrome trace (reference parity: Engine's
        trace_static.json export, models/engine.py:153-179 — behavior
        only). Requires a prior serve()/decode_once() so the graph is
        captured."""
        self._ensure_graph()
        torch.cuda.synchronize()
        with torch.profiler.profile(
                activities=[torch.profiler.ProfilerActivity.CUDA]) as prof:
            for _ in range(steps):
                self.graph.replay()
            torch.cuda.synchronize()
        if trace_path:
            prof.export_chrome_trace(trace_path)
        rows = {}
        for e in prof.key_averages():
            t = getattr(e, "self_device_time_total", 0) or 0
            if t > 0:
                rows[e.key] = {"us_per_step": t / steps, "calls": e.count}
        return rows

    def serve(self, input_ids: torch.Tensor, gen_len: int) -> torch.Tensor:
        """input_ids: [B, S] prompt -> [B, gen_len] generated (greedy)."""
        b, s = input_ids.shape
        assert b == self.batch and s + gen_len <= self.max_len
        self.kv.reset()
        first = self.model.prefill(input_ids, self.kv)
        out = [first]
        tok = first
        if self.use_graph:
            self._ensure_graph()
            self._token_buf.copy_(tok)
            for _ in range(gen_len - 1):
                self.graph.replay()
                out.append(self._token_buf.clone())
            # KV offset advanced inside the graph; nothing to fix up
        else:
            for _ in range(gen_len - 1):
                tok = self.decode_once(tok)
                out.append(tok.clone())
        return torch.stack(out, dim=1)
