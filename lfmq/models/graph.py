"""hipGraph-captured decode.

The eager decode step is ~260 kernel launches; at ~1.2-3 µs of
launch/gap each that overhead rivals the weight-stream time.  The whole
step — embedding, all layers, lm_head, greedy sampling and the decode
state advance — is captured ONCE into a hipGraph and replayed per token
with zero host work: position/seq-len live in a device state tensor
{pos0, T_after, step} that the in-graph advance_record kernel bumps, the
sampled token is fed back into the graph's own static input buffer, and
every generated token lands in a device-side ring read back once at the
end.  torch.cuda.CUDAGraph is hipGraph on ROCm.
"""
from __future__ import annotations

from typing import Optional

import torch

from ..ops import functional as F
from ..ops import hip_ext


class GraphDecoder:
    def __init__(self, model, batch: int, max_seq: int,
                 max_new_tokens: int = 4096):
        self.model = model
        self.device = next(model.parameters()).device
        assert self.device.type == "cuda", "GraphDecoder is GPU-only"
        self.batch = batch
        self.max_seq = max_seq
        self.cache = model.make_cache(batch, max_seq=max_seq, device=self.device)
        self.state = torch.zeros(3, dtype=torch.int32, device=self.device)
        self.tok = torch.zeros(batch, dtype=torch.int64, device=self.device)
        self.ring = torch.zeros(batch, max_new_tokens, dtype=torch.int64,
                                device=self.device)
        self.step_count = 0
        self.graph: Optional[torch.cuda.CUDAGraph] = None

    @torch.no_grad()
    def _step(self):
        logits = self.model.forward_decode_step(
            self.tok.unsqueeze(1), self.cache, self.state, self.max_seq)
        sampled = hip_ext().greedy_sample(logits[:, -1].contiguous())
        hip_ext().advance_record(self.state, self.tok, sampled, self.ring)

    @torch.no_grad()
    def capture(self):
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):  # warmup on a side stream (allocator prep)
                self._step()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        # reset whatever the warmup advanced
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self._step()

    @torch.no_grad()
    def prefill(self, input_ids: torch.Tensor) -> None:
        """Eager prefill + set up the device state for decode."""
        b, L = input_ids.shape
        assert b == self.batch
        self.cache.reset()
        logits = self.model(input_ids.to(self.device), self.cache)
        first = F.greedy_sample(logits[:, -1].contiguous())
        self.tok.copy_(first)
        self.state.copy_(torch.tensor([L, L + 1, 0], dtype=torch.int32))
        self.ring.zero_()
        self.step_count = 0

    @torch.no_grad()
    def decode(self, n_tokens: int) -> torch.Tensor:
        """Replay the captured step n_tokens times; returns
        [batch, n_tokens] starting with the first post-prefill token
        (each replay records the token it consumed and feeds the newly
        sampled one back into the graph's input buffer)."""
        if self.step_count + n_tokens > self.ring.shape[1]:
            raise RuntimeError(
                f"token ring exhausted: {self.step_count} + {n_tokens} > "
                f"{self.ring.shape[1]} (raise max_new_tokens at construction)")
        if self.cache.seq_len + n_tokens > self.max_seq:
            raise RuntimeError(
                f"KV cache exhausted: {self.cache.seq_len} + {n_tokens} > "
                f"{self.max_seq} (the in-graph append would write past the "
                f"cache; raise max_seq at construction)")
        if self.graph is None:
            # capture AFTER a real prefill so shapes/addresses are final
            saved = (self.state.clone(), self.tok.clone(),
                     self.cache.seq_len)
            self.capture()
            self.state.copy_(saved[0])
            self.tok.copy_(saved[1])
            self.cache.seq_len = saved[2]
        start = self.step_count
        for _ in range(n_tokens):
            self.graph.replay()
        self.step_count += n_tokens
        self.cache.seq_len += n_tokens
        eng = getattr(self.model, "_engine", None)
        if eng is not None:
            eng.check_health()  # surfaces engine spin give-up codes
        return self.ring[:, start:start + n_tokens].clone()
