"""hipGraph-captured decode engine.

The eager decode step is launch-bound (~200 kernel dispatches per token);
capturing one step in a hipGraph (``torch.cuda.CUDAGraph`` on ROCm) and
replaying it per token removes that overhead — the MI355X answer to the
reference's traced/compiled token-generation model (reference
trace/model_builder + nxd_model bucket executor).

The step is made capturable by driving every position-dependent piece of
the model from DEVICE tensors (see ``LlamaAttention._decode_step`` and
``KVCache.update`` tensor-pos paths): between replays only the contents
of ``step_in`` (next token) and ``pos_t`` change.
"""

from typing import List, Optional

import torch

from .kv_cache import KVCache


class GraphDecoder:
    """Wraps a causal-LM for single-token decode via graph replay.

    Usage:
        logits = model(prompt, kv_caches=caches, pos_offset=0)   # prefill
        dec = GraphDecoder(model, caches, start_pos=prompt_len)
        logits = dec.step(next_tok)   # (B, 1, V/tp); repeat per token

    Multi-step capture (``steps_per_capture > 1``): the captured graph runs
    N whole decode iterations INCLUDING the greedy next-token selection
    (device-side feedback through ``step_in``), so one replay emits N
    tokens — amortizing the inter-kernel replay gaps and the per-token
    host loop:
        toks = dec.step_block(next_tok)   # (B, N) token ids
    """

    def __init__(self, model, kv_caches: List[KVCache], start_pos: int,
                 batch: int, device=None, steps_per_capture: int = 1):
        self.model = model
        self.caches = kv_caches
        device = device or torch.device("cuda",
                                        torch.cuda.current_device())
        self.steps_per_capture = steps_per_capture
        self.step_in = torch.zeros(batch, 1, dtype=torch.long, device=device)
        self.pos_t = torch.tensor([start_pos], dtype=torch.long,
                                  device=device)
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.logits: Optional[torch.Tensor] = None
        self.tokens_out: Optional[torch.Tensor] = None

    def _eager(self):
        # no_grad so the model's inference-only fused kernels (add_rmsnorm,
        # fused-QKV single GEMM) are active in the CAPTURED graph too
        with torch.no_grad():
            return self.model(self.step_in, kv_caches=self.caches,
                              pos_offset=self.pos_t)

    def _greedy_next(self, logits):
        """Device-side greedy token over (possibly vocab-sharded) logits —
        graph-capturable (argmax + the TP gather are both capturable)."""
        from ..operators import argmax as dist_argmax

        return dist_argmax(logits[:, -1, :], dim=-1, gather_dim=-1)

    def _eager_block(self):
        """steps_per_capture whole decode iterations with device-side
        greedy feedback; returns (tokens (B, N), last logits)."""
        with torch.no_grad():
            toks = []
            for _ in range(self.steps_per_capture):
                logits = self.model(self.step_in, kv_caches=self.caches,
                                    pos_offset=self.pos_t)
                nxt = self._greedy_next(logits)
                self.step_in.copy_(nxt.view(-1, 1))
                self.pos_t += 1
                toks.append(nxt)
            return torch.stack(toks, dim=1), logits

    def capture(self) -> bool:
        """Capture one decode step; returns False (eager fallback) if
        capture fails (e.g. a non-capturable collective)."""
        try:
            # TunableOp TUNING mode launches candidate kernels mid-step —
            # incompatible with stream capture (observed GPU memory
            # faults).  Replaying already-tuned selections is fine.
            tun = torch.cuda.tunable
            if tun.is_enabled() and tun.tuning_is_enabled():
                return False
        except Exception:
            pass
        try:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):  # warmup allocations off-graph
                for _ in range(2):
                    self._eager()
            torch.cuda.current_stream().wait_stream(side)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self.logits = self._eager()
            self.graph = g
            return True
        except Exception as e:
            from ..utils.logger import get_logger

            get_logger(__name__).warning(
                "hipGraph capture failed (%s: %s) — decoding eagerly",
                type(e).__name__, e)
            self.graph = None
            return False

    @torch.no_grad()
    def step(self, next_tok: torch.Tensor) -> torch.Tensor:
        """next_tok (B,) -> logits (B, 1, V/tp); advances the position."""
        self.step_in.copy_(next_tok.view(-1, 1))
        if self.graph is not None:
            self.graph.replay()
            out = self.logits
        else:
            out = self._eager()
        self.pos_t += 1
        return out

    def capture_block(self) -> bool:
        """Capture a ``steps_per_capture``-token decode graph with the
        greedy feedback inside; returns False on capture failure (callers
        fall back to per-step decoding)."""
        if self.steps_per_capture <= 1:
            return self.capture()
        try:
            pos0 = self.pos_t.clone()
            in0 = self.step_in.clone()
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):  # warmup allocations off-graph
                for _ in range(2):
                    self.pos_t.copy_(pos0)
                    self.step_in.copy_(in0)
                    self._eager_block()
            torch.cuda.current_stream().wait_stream(side)
            self.pos_t.copy_(pos0)
            self.step_in.copy_(in0)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self.tokens_out, self.logits = self._eager_block()
            self.graph = g
            # capture itself ran the block once: rewind the position
            self.pos_t.copy_(pos0)
            return True
        except Exception as e:
            from ..utils.logger import get_logger

            get_logger(__name__).warning(
                "multi-step hipGraph capture failed (%s: %s) — per-step",
                type(e).__name__, e)
            self.graph = None
            return False

    @torch.no_grad()
    def step_block(self, next_tok: torch.Tensor) -> torch.Tensor:
        """next_tok (B,) -> (B, steps_per_capture) greedily decoded token
        ids (one graph replay; position advances inside the graph)."""
        self.step_in.copy_(next_tok.view(-1, 1))
        if self.graph is not None:
            self.graph.replay()
            return self.tokens_out.clone()
        toks, _ = self._eager_block()
        return toks
