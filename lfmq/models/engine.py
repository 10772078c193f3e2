"""Persistent weight-streaming decode engine (host side).

Builds the per-CU weight streams consumed by csrc/engine.hip and drives
the one-launch-per-step decode path.  The engine replaces the 6-kernel/
layer launched chain for b=1 s=1 TP=1 W4 Llama decode; everything else
(prefill, batch>1, TP, quantizers, other archs) keeps the launched
kernels.  Kill switch: LFMQ_ENGINE=0.

Stream layout (must mirror csrc/engine.hip record walking EXACTLY):
  for layer in layers:
    for op in [qkv_proj, o_proj, gate_up_proj, down_proj]:
      for each of this CU's records (columns; gate_up interleaves
          [gate_j, up_j, ...]):
        [k/2 B qweight row bytes][G*2 B bf16 scales][ceil(G/2) B packed
         int4 zeros][pad to 16 B]
  - CU c owns columns [c*n/256, (c+1)*n/256) of every op.
  - a record that would straddle a ring-wrap boundary (nslot_ring *
    16384 bytes) is pushed to the next boundary (padding streamed but
    never read).
  - every CU's stream is byte-identical in SHAPE (offsets), so one
    offset walk serves all 256.
"""
from __future__ import annotations

import os
from typing import List, Optional

import torch

from ..ops import functional as F
from ..ops import hip_ext

SLOT = 16384
NCU = 256
LDS_BYTES = 160 * 1024
AT = 256     # attention chunk (engine kernel constant)
NCONS = 3    # consumer waves per CU (engine kernel constant)


def _rec_bytes(k: int, G: int) -> int:
    """Stream-record bytes, format v2: lane-major nibble dwords + fp32
    (s, s*(z+128)) pairs (csrc/engine_common.h record_dot)."""
    return (k // 2 + G * 8 + 15) & ~15


def _is_pow2(x: int) -> bool:
    return x > 0 and (x & (x - 1)) == 0


def engine_supported(model, batch: int) -> bool:
    """Static conditions for the persistent engine (8B/70B shapes)."""
    from .llama import LlamaForCausalLM
    from .qlinear import QLinear
    # OPT-IN this round: the engine is correct (bit-deterministic, parity
    # within launched-variant noise — tests/test_gpu_engine.py) but its
    # last full-8B measurement (183 tok/s, single-record dots) trails the
    # launched path's 522; the record-pair ILP rework (0.849 -> 0.481
    # us/record in the ubench) landed after that measurement and projects
    # ~250-300 — still short, so the default stays on the launched
    # kernels (docs/persistent_decode_design.md status section).
    if os.environ.get("LFMQ_ENGINE", "0") != "1":
        return False
    if not isinstance(model, LlamaForCausalLM) or batch != 1:
        return False
    from .. import dist as lfmq_dist
    if lfmq_dist.get_world_size() > 1:
        return False
    cfg = model.cfg
    if cfg.head_dim != 128:
        return False
    nqkv = (cfg.n_heads + 2 * cfg.n_kv_heads) * cfg.head_dim
    # k dims % 2048: record_dot runs branch-free 4x512-value chunks
    if (cfg.hidden_size % 2048 or cfg.intermediate_size % 2048 or nqkv % 512
            or (cfg.n_heads * cfg.head_dim) % 2048):
        return False
    g = None
    for layer in model.layers:
        if not (getattr(layer, "_in_norm_folded", False)
                and getattr(layer, "_post_norm_folded", False)):
            return False
        for proj in (layer.self_attn.qkv_proj, layer.self_attn.o_proj,
                     layer.mlp.gate_up_proj, layer.mlp.down_proj):
            if not isinstance(proj, QLinear) or proj.bits != 4:
                return False
            if proj.bias is not None or hasattr(proj, "perm"):
                return False
            if g is None:
                g = proj.group_size
            if proj.group_size != g or not _is_pow2(g) or g < 32:
                return False
    return True


def _ring_slots(cfg) -> int:
    """Deepest 16 KiB ring that fits LDS beside xbuf/union/scratch/ctl."""
    union = max(cfg.intermediate_size * 2,
                (cfg.n_heads + 2 * cfg.n_kv_heads) * cfg.head_dim * 2,
                cfg.n_heads * cfg.head_dim * 2)
    ascr = max(NCONS * cfg.head_dim * 4 + NCONS * AT * 4,
               cfg.hidden_size * 2)
    fixed = cfg.hidden_size * 2 + union + ((ascr + 63) & ~63) + 256
    return min(7, (LDS_BYTES - fixed) // SLOT)


class DecodeEngine:
    """Bound to one (model, cache) pair; rebuilt if either changes."""

    def __init__(self, model, cache, max_seq: int):
        cfg = model.cfg
        self.model = model
        self.cache = cache
        self.max_seq = max_seq
        dev = next(model.parameters()).device
        self.device = dev
        self.nslot = _ring_slots(cfg)
        assert self.nslot >= 3, "model too wide for the engine LDS budget"
        self.g = model.layers[0].self_attn.qkv_proj.group_size
        self.log2g = self.g.bit_length() - 1
        self.nqkv = (cfg.n_heads + 2 * cfg.n_kv_heads) * cfg.head_dim
        self.stream, self.per_cu = self._build_stream(model, dev)
        hq, d = cfg.n_heads, cfg.head_dim
        hid, inter = cfg.hidden_size, cfg.intermediate_size
        tmax = cache.k_all.shape[3]
        assert cache.k_all.shape[1] == 1, "engine is b=1"
        nch_max = (tmax + AT - 1) // AT
        i64 = dict(dtype=torch.int64, device=dev)
        self.g_qkv = torch.zeros(self.nqkv // 2, **i64)
        self.g_attn = torch.zeros(hq * d // 2, **i64)
        self.g_o = torch.zeros(hid // 2, **i64)
        self.g_h = torch.zeros(inter // 2, **i64)
        self.g_down = torch.zeros(hid // 2, **i64)
        self.g_scr = torch.zeros(hq * nch_max * (d + 2), **i64)
        self.tmo = torch.zeros(8, dtype=torch.int32, device=dev)
        self.xout = torch.zeros(1, 1, hid, dtype=torch.bfloat16, device=dev)
        self.inv_freq = F._inv_freq(d, cfg.rope_theta, dev)
        self.attn_scale = 1.0 / (d ** 0.5)
        self.debug = int(os.environ.get("LFMQ_ENGINE_DEBUG", "0"))
        self.prof: Optional[torch.Tensor] = None
        if os.environ.get("LFMQ_ENGINE_PROF"):
            n_slots = (self.per_cu + SLOT - 1) // SLOT
            self.prof = torch.zeros(cfg.n_layers * 8 + n_slots,
                                    dtype=torch.int64, device=dev)

    # ------------------------------------------------------------ build
    @torch.no_grad()
    def _build_stream(self, model, dev):
        cfg = model.cfg
        ring = self.nslot * SLOT

        def op_records(q, interleave_half: bool):
            """[NCU, npc, rec_b] uint8 record block for one QLinear
            (format v2: lane-major nibbles, fp32 (s, sz) pairs)."""
            n, k = q.out_features, q.in_features
            G = k // self.g
            rb = _rec_bytes(k, G)
            iters = k // 512
            # nibbles: dword for (iter i, lane l) at (l*iters + i):
            # transpose the row's [iters, 64] dword grid to [64, iters]
            qw = q.qweight.reshape(n, iters, 64).transpose(1, 2).contiguous()
            nib = qw.view(torch.uint8).reshape(n, k // 2)
            # fp32 (s, s*(z+128)) pairs — sz computed exactly as the v1
            # kernel did (bf16 s -> fp32, times the int zero bias)
            s = q.scales.float()
            gidx = torch.arange(G, device=q.qzeros.device)
            z = (q.qzeros[:, gidx // 8] >> (4 * (gidx % 8))) & 0xF
            sz = s * (z.float() + 128.0)
            pairs = torch.stack([s, sz], dim=-1).reshape(n, 2 * G)
            pb = pairs.view(torch.uint8).reshape(n, G * 8)
            rec = torch.zeros(n, rb, dtype=torch.uint8, device=dev)
            rec[:, :k // 2] = nib
            rec[:, k // 2:k // 2 + G * 8] = pb
            if interleave_half:
                # gate_up: rows [g_0..g_{I-1}, u_0..u_{I-1}] -> per-CU
                # interleaved [g_j, u_j, ...]
                inter = n // 2
                m = inter // NCU
                # for cu c, slot 2j -> gate row c*m+j, 2j+1 -> up row inter+c*m+j
                c = torch.arange(NCU).repeat_interleave(2 * m)
                j = torch.arange(2 * m).repeat(NCU)
                gate = c * m + j // 2
                idx = torch.where(j % 2 == 0, gate, inter + gate)
                rec = rec[idx.to(dev)]
                return rec.reshape(NCU, 2 * m, rb)
            npc = n // NCU
            return rec.reshape(NCU, npc, rb)

        # offsets walk (identical for every CU)
        blocks = []  # (tensor [NCU, npc, rb], rb)
        for layer in model.layers:
            blocks.append(op_records(layer.self_attn.qkv_proj, False))
            blocks.append(op_records(layer.self_attn.o_proj, False))
            blocks.append(op_records(layer.mlp.gate_up_proj, True))
            blocks.append(op_records(layer.mlp.down_proj, False))

        offs: List[List[int]] = []
        cur = 0
        for blk in blocks:
            rb = blk.shape[2]
            bo = []
            for _ in range(blk.shape[1]):
                if cur % ring + rb > ring:
                    cur += ring - cur % ring
                bo.append(cur)
                cur += rb
            offs.append(bo)
        per_cu = (cur + SLOT - 1) // SLOT * SLOT
        stream = torch.zeros(NCU, per_cu, dtype=torch.uint8, device=dev)
        for blk, bo in zip(blocks, offs):
            rb = blk.shape[2]
            r = 0
            while r < len(bo):  # copy contiguous runs in one slice
                r2 = r
                while r2 + 1 < len(bo) and bo[r2 + 1] == bo[r2] + rb:
                    r2 += 1
                n = r2 - r + 1
                stream[:, bo[r]:bo[r] + n * rb] = \
                    blk[:, r:r + n].reshape(NCU, n * rb)
                r = r2 + 1
        del blocks
        return stream.reshape(-1), per_cu

    # ------------------------------------------------------------- step
    def step(self, input_ids: torch.Tensor, state: torch.Tensor):
        """One decode step's layer stack: [1,1] ids -> [1,1,hidden]
        (final-norm applied; caller runs lm_head)."""
        cfg = self.model.cfg
        # zero every granule word before every launch (replayed memset
        # nodes under graph capture — Guideline 16 state re-init)
        self.g_qkv.zero_()
        self.g_attn.zero_()
        self.g_o.zero_()
        self.g_h.zero_()
        self.g_down.zero_()
        self.g_scr.zero_()
        hip_ext().engine_step(
            self.stream, self.per_cu, self.nslot, cfg.n_layers,
            cfg.hidden_size, cfg.intermediate_size, self.nqkv, cfg.n_heads,
            cfg.n_kv_heads, self.log2g, cfg.rms_eps, self.attn_scale,
            self.inv_freq, self.cache.k_all, self.cache.v_all, state,
            input_ids.reshape(-1), self.model.embed_tokens.weight,
            self.model.norm.weight, self.xout, self.g_qkv, self.g_attn,
            self.g_o, self.g_h, self.g_down, self.g_scr, self.tmo, self.prof,
            self.debug)
        return self.xout

    def check_health(self):
        """Host-side give-up check (cheap sync; call between batches)."""
        code = int(self.tmo[0].item())
        if code != 0:
            raise RuntimeError(f"decode engine spin give-up code 0x{code:x}")
