"""hipGraph-captured decode steps (torch.cuda.CUDAGraph == hipGraph on ROCm).

The decode forward launches ~85 small kernels per step; eager dispatch
costs ~2 ms/step of host gap (profiles/r01_decode_profile_v1.md). Graphs
are captured lazily per (batch-bucket, seqlen-bucket) with static input
buffers and a shared memory pool, replayed with a single launch.
Sampling stays outside the graph (host-side seeds/temperatures vary).
"""
from __future__ import annotations

from typing import Dict, Tuple

import torch

BATCH_BUCKETS = [1, 2, 4, 8, 16, 32, 48, 64, 96, 128, 192, 256,
                 384, 512]
MIN_LEN_BUCKET = 512


class CUDAGraphRunner:
    def __init__(self, model, kv_caches, workspace, max_model_len: int,
                 block_size: int, max_batch: int, device):
        self.model = model
        self.kv_caches = kv_caches
        self.workspace = workspace
        self.max_model_len = max_model_len
        self.block_size = block_size
        self.max_batch = max_batch
        self.device = device
        self.max_blocks = max_model_len // block_size + 1
        self.pool = None
        self.graphs: Dict[Tuple[int, int], dict] = {}

        B = max_batch
        self.in_ids = torch.zeros(B, dtype=torch.int64, device=device)
        self.in_pos = torch.zeros(B, dtype=torch.int64, device=device)
        self.in_slots = torch.full((B,), -1, dtype=torch.int64, device=device)
        self.in_bt = torch.zeros(B, self.max_blocks, dtype=torch.int32,
                                 device=device)
        self.in_lens = torch.ones(B, dtype=torch.int32, device=device)
        # pinned host staging (one DMA per field per step)
        self.h_ids = torch.zeros(B, dtype=torch.int64, pin_memory=True)
        self.h_pos = torch.zeros(B, dtype=torch.int64, pin_memory=True)
        self.h_slots = torch.zeros(B, dtype=torch.int64, pin_memory=True)
        self.h_lens = torch.ones(B, dtype=torch.int32, pin_memory=True)
        self.h_bt = torch.zeros(B, self.max_blocks, dtype=torch.int32,
                                pin_memory=True)

    def batch_bucket(self, n: int) -> int:
        for b in BATCH_BUCKETS:
            if b >= n:
                return min(b, self.max_batch)
        return self.max_batch

    def len_bucket(self, max_len: int) -> int:
        b = MIN_LEN_BUCKET
        while b < max_len:
            b *= 2
        return min(b, self.max_model_len)

    def _capture(self, nb: int, lb: int) -> dict:
        from helix_amd.models.llama import DecodeMeta
        meta = DecodeMeta(
            block_tables=self.in_bt[:nb],
            seq_lens=self.in_lens[:nb],
            slot_mapping=self.in_slots[:nb],
            positions=self.in_pos[:nb],
            max_len=lb,
            workspace=self.workspace)
        ids = self.in_ids[:nb]
        # Warm up on a side stream (allocator state, lazy inits).
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                hidden = self.model(ids, self.kv_caches, meta)
                logits = self.model.compute_logits(hidden)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

        g = torch.cuda.CUDAGraph()
        ctx = (torch.cuda.graph(g, pool=self.pool) if self.pool is not None
               else torch.cuda.graph(g))
        with ctx:
            hidden = self.model(ids, self.kv_caches, meta)
            logits = self.model.compute_logits(hidden)
        if self.pool is None:
            self.pool = g.pool()
        return {"graph": g, "logits": logits}

    def run(self, input_ids, positions, slots, block_tables, seq_lens,
            max_len: int) -> torch.Tensor:
        """All args numpy arrays (or lists); returns logits[:B]."""
        B = len(input_ids)
        nb = self.batch_bucket(B)
        lb = self.len_bucket(max_len)
        key = (nb, lb)
        # Stage inputs into pinned host buffers then one async DMA per
        # field (pad rows are inert: slot -1 => no KV write; seq_len 1 =>
        # one garbage token read).
        self.h_ids[:nb] = 0
        self.h_pos[:nb] = 0
        self.h_slots[:nb] = -1
        self.h_lens[:nb] = 1
        self.h_bt[:nb].zero_()
        self.h_ids[:B] = torch.as_tensor(input_ids, dtype=torch.int64)
        self.h_pos[:B] = torch.as_tensor(positions, dtype=torch.int64)
        self.h_slots[:B] = torch.as_tensor(slots, dtype=torch.int64)
        self.h_lens[:B] = torch.as_tensor(seq_lens, dtype=torch.int32)
        bt_t = torch.as_tensor(block_tables)
        w = bt_t.shape[1]
        self.h_bt[:B, :w] = bt_t
        self.in_ids[:nb].copy_(self.h_ids[:nb], non_blocking=True)
        self.in_pos[:nb].copy_(self.h_pos[:nb], non_blocking=True)
        self.in_slots[:nb].copy_(self.h_slots[:nb], non_blocking=True)
        self.in_lens[:nb].copy_(self.h_lens[:nb], non_blocking=True)
        self.in_bt[:nb].copy_(self.h_bt[:nb], non_blocking=True)

        if key not in self.graphs:
            self.graphs[key] = self._capture(nb, lb)
        entry = self.graphs[key]
        entry["graph"].replay()
        return entry["logits"][:B]

    def all_len_buckets(self):
        """Every seqlen bucket up to max_model_len (512, 1024, ...)."""
        out = []
        b = MIN_LEN_BUCKET
        while True:
            out.append(min(b, self.max_model_len))
            if b >= self.max_model_len:
                break
            b *= 2
        return out

    def warmup(self, len_buckets=None):
        """Eagerly capture ALL (batch, seqlen) buckets so serving never
        pays multi-second capture latency mid-request — including long
        contexts (cold-start cost moves to model load; bound it by
        lowering the spec's max_model_len if load time matters more)."""
        if len_buckets is None:
            len_buckets = self.all_len_buckets()
        self.in_ids.zero_()
        self.in_pos.zero_()
        self.in_slots.fill_(-1)
        self.in_lens.fill_(1)
        self.in_bt.zero_()
        for lb in len_buckets:
            lb = min(lb, self.max_model_len)
            for nb in BATCH_BUCKETS:
                if nb > self.max_batch:
                    break
                key = (nb, lb)
                if key not in self.graphs:
                    self.graphs[key] = self._capture(nb, lb)
