"""hipGraph-captured training microbatch — streams & graphs, not a tracing
compiler (the north-star runtime design; no Triton / torch.compile).

Captures ONE fixed-shape microbatch forward+backward into a hipGraph
(torch.cuda.CUDAGraph is hipGraph on ROCm) and replays it per microbatch:
removes the per-kernel launch overhead of the ~80-kernel microbatch body.

Measured at MPT-125M/gb256/mb32: graphs 601-606k vs eager 611-616k
tokens/s, both with fp32+autocast and with bf16 master-weight mode — the
~10-16 us replay floor over 8 microbatches plus graph-pool memory effects
eat the launch-overhead win at this kernel count. Opt-in
(llm_config.use_hip_graphs); expected to pay off for smaller models or
larger grad-accum degrees.

Contract:
* input shape is fixed (microbatch_size x seq_len) — the bench/fed training
  path pads nothing and always ticks full microbatches;
* gradients ACCUMULATE into the captured .grad tensors — callers must zero
  with ``set_to_none=False`` (zero_()), never free them;
* the optimizer step and scheduler run OUTSIDE the graph (dynamic lr);
* weight VALUES may change between replays (the fed round copies new
  globals into the same parameter tensors — graph-safe).
"""

from __future__ import annotations

import torch


class GraphedMicrobatch:
    def __init__(self, model, autocast_ctx_factory, loss_div: float,
                 microbatch: int, seq_len: int, device):
        self.model = model
        self.device = device
        self.static_ids = torch.zeros(
            (microbatch, seq_len), dtype=torch.long, device=device
        )
        self.loss_div = loss_div
        self._graph = None
        self._static_loss = None
        self._autocast = autocast_ctx_factory

    def _capture(self):
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._zero_grads()
                with self._autocast():
                    out = self.model(self.static_ids, labels=self.static_ids)
                    loss = out["loss"] / self.loss_div
                    loss.backward()
        torch.cuda.current_stream().wait_stream(s)
        self._zero_grads()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            with self._autocast():
                out = self.model(self.static_ids, labels=self.static_ids)
                loss = out["loss"] / self.loss_div
                loss.backward()
            self._static_loss = loss.detach()
        self._graph = g

    def _zero_grads(self):
        for p in self.model.parameters():
            if p.grad is not None:
                p.grad.zero_()

    def run(self, input_ids: torch.Tensor) -> torch.Tensor:
        """Replay the captured microbatch; returns the (device) loss."""
        if self._graph is None:
            self._capture()
        self.static_ids.copy_(input_ids, non_blocking=True)
        self._graph.replay()
        return self._static_loss
