"""Calibration data pipeline: fixed-seed synthetic token streams.

No network anywhere in this environment, so calibration
runs on seeded synthetic token sequences — matching the BASELINE.json
protocol ("on synthetic data / random-init weights").  Batches stream
through the model block-by-block (sequential per-block calibration keeps
peak memory at one decoder block, which is what lets a 70B quantize fit).
"""
from __future__ import annotations

from typing import Iterator, List

import torch
import torch.nn as nn


class CalibStream:
    def __init__(self, vocab_size: int, seq_len: int = 512, n_seqs: int = 16,
                 seed: int = 1234, batch: int = 4):
        self.vocab_size = vocab_size
        self.seq_len = seq_len
        self.n_seqs = n_seqs
        self.seed = seed
        self.batch = batch

    def batches(self) -> Iterator[torch.Tensor]:
        g = torch.Generator().manual_seed(self.seed)
        done = 0
        while done < self.n_seqs:
            b = min(self.batch, self.n_seqs - done)
            yield torch.randint(0, self.vocab_size, (b, self.seq_len), generator=g)
            done += b


class InputCatcher:
    """Forward hooks that record each target Linear's layer inputs.

    Used by GPTQ (feeds ops.hessian_accum per batch) and AWQ (feeds
    per-channel absmax stats).  consume=None just stores [ntok, k] tensors.
    """

    def __init__(self, modules: dict[str, nn.Module], consume=None):
        self.modules = modules
        self.consume = consume
        self.store: dict[str, List[torch.Tensor]] = {name: [] for name in modules}
        self._handles = []

    def __enter__(self):
        for name, mod in self.modules.items():
            def hook(m, inp, out, _name=name):
                x = inp[0].detach()
                x = x.reshape(-1, x.shape[-1])
                if self.consume is not None:
                    self.consume(_name, x)
                else:
                    self.store[_name].append(x)
            self._handles.append(mod.register_forward_hook(hook))
        return self

    def __exit__(self, *exc):
        for h in self._handles:
            h.remove()
        self._handles.clear()
        return False
