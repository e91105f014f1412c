"""LLMEngine — continuous-batching inference engine for MI355X.

Replaces the vLLM container the reference dispatches to
(SURVEY.md §3.3 hot loop: "continuous batching, paged attention,
sampling — all delegated"). Prefill and decode interleave: each step()
either admits waiting sequences with one varlen prefill or advances all
running sequences one decode token over the paged KV cache.
"""
from __future__ import annotations

import time
import zlib
from dataclasses import dataclass, field
from enum import Enum
from typing import Callable, Dict, List, Optional

import numpy as np
import torch

from helix_amd import ops
from helix_amd.engine.kv_cache import KVCache, chain_hash
from helix_amd.engine.sampling_params import SamplingParams
from helix_amd.models.llama import (DecodeMeta, LlamaConfig, LlamaForCausalLM,
                                    PrefillMeta, PRESETS)

_SEED_MIX = 0x9E3779B97F4A7C15


class SeqStatus(Enum):
    WAITING = "waiting"
    RUNNING = "running"
    FINISHED = "finished"
    CANCELLED = "cancelled"


@dataclass
class Sequence:
    seq_id: str
    prompt_ids: List[int]
    params: SamplingParams
    output_ids: List[int] = field(default_factory=list)
    status: SeqStatus = SeqStatus.WAITING
    block_table: List[int] = field(default_factory=list)
    row: int = -1                          # persistent block-table row slot
    cached_prefix: int = 0                 # tokens served from prefix cache
    block_hashes: List[bytes] = field(default_factory=list)
    arrival_time: float = field(default_factory=time.monotonic)
    first_token_time: Optional[float] = None
    finish_reason: Optional[str] = None
    logprobs: List[dict] = field(default_factory=list)  # when requested
    pen_init: bool = False       # GPU penalty tables primed for this row
    json_mask: object = None     # JSONByteMask when params.json_mode
    # streaming callback: fn(seq, new_token_id, finished)
    on_token: Optional[Callable] = None

    @property
    def total_len(self) -> int:
        return len(self.prompt_ids) + len(self.output_ids)


@dataclass
class EngineConfig:
    model: str = "llama3-8b"
    block_size: int = 16
    max_num_seqs: int = 64
    max_model_len: int = 8192
    max_prefill_tokens: int = 8192
    kv_cache_blocks: Optional[int] = None     # None => size from free HBM
    gpu_memory_utilization: float = 0.90
    eos_token_id: int = 2
    seed: int = 0
    enforce_eager: bool = False               # disable hipGraph capture
    enable_prefix_caching: bool = True
    quantization: Optional[str] = None        # None | "fp8" (e4m3 W8A8)
    kv_cache_dtype: str = "bf16"              # "bf16" | "fp8" (e4m3 KV)


class LLMEngine:
    def __init__(self, cfg: EngineConfig, device: str = "cuda",
                 model: Optional[LlamaForCausalLM] = None,
                 tp_size: int = 1, tp_rank: int = 0,
                 tp_group=None):
        self.cfg = cfg
        self.device = torch.device(device)
        self.model_cfg: LlamaConfig = PRESETS[cfg.model]
        self.tp_size = tp_size
        self.tp_rank = tp_rank
        self.tp_group = tp_group
        if model is None:
            dtype = torch.bfloat16
            model = LlamaForCausalLM(self.model_cfg, tp_size, tp_rank)
            model = model.to(dtype).to(self.device)
            model.init_random(cfg.seed)
        if cfg.quantization == "fp8":
            from helix_amd.models.quant import quantize_model_fp8
            quantize_model_fp8(model)
        self.model = model

        nkv = self.model_cfg.num_kv_heads // tp_size
        num_blocks = cfg.kv_cache_blocks
        if num_blocks is None:
            num_blocks = self._auto_kv_blocks(nkv)
        kv_dtype = torch.uint8 if cfg.kv_cache_dtype == "fp8" \
            else torch.bfloat16
        self.kv = KVCache(self.model_cfg.num_layers, nkv,
                          self.model_cfg.head_dim, cfg.block_size, num_blocks,
                          self.device, dtype=kv_dtype)
        self.waiting: List[Sequence] = []
        self.running: List[Sequence] = []
        self.seqs: Dict[str, Sequence] = {}
        self.steps = 0
        if self.device.type == "cuda":
            self.decode_ws = ops.decode_workspace(
                cfg.max_num_seqs, self.model_cfg.num_heads // tp_size,
                self.model_cfg.head_dim, cfg.max_model_len, self.device)
        else:
            self.decode_ws = None
        # GPU penalty tables (lazy): [max_num_seqs, V] output counts +
        # prompt-seen bitmap, read by the fused sampling kernel
        self._pen_counts: Optional[torch.Tensor] = None
        self._pen_seen: Optional[torch.Tensor] = None
        # Persistent sampling-parameter staging: one pinned->device copy
        # per dtype per step instead of 8 small allocations + H2D syncs.
        if self.device.type == "cuda":
            R = cfg.max_num_seqs
            # rows: 0=temp 1=top_p 2=rep 3=pres 4=freq
            self._samp_hf = torch.empty(5, R, dtype=torch.float32,
                                        pin_memory=True)
            self._samp_hs = torch.empty(R, dtype=torch.int64,
                                        pin_memory=True)
            # rows: 0=top_k 1=row_map
            self._samp_hi = torch.empty(2, R, dtype=torch.int32,
                                        pin_memory=True)
            self._samp_df = torch.empty(5, R, dtype=torch.float32,
                                        device=self.device)
            self._samp_ds = torch.empty(R, dtype=torch.int64,
                                        device=self.device)
            self._samp_di = torch.empty(2, R, dtype=torch.int32,
                                        device=self.device)
        # Persistent host-side batch state: per-seq block-table rows and
        # scratch arrays. Rebuilding these as Python-list -> torch.tensor
        # per step cost ~4 ms at B=512 (512 small tensor constructions);
        # numpy row updates are O(changed entries) instead.
        mb = cfg.max_model_len // cfg.block_size + 2
        self._bt_np = np.zeros((cfg.max_num_seqs, mb), dtype=np.int32)
        self._free_rows = list(range(cfg.max_num_seqs - 1, -1, -1))
        self._ids_np = np.zeros(cfg.max_num_seqs, dtype=np.int64)
        self._pos_np = np.zeros(cfg.max_num_seqs, dtype=np.int64)
        self._slots_np = np.zeros(cfg.max_num_seqs, dtype=np.int64)
        self._lens_np = np.zeros(cfg.max_num_seqs, dtype=np.int32)
        self._rows_np = np.zeros(cfg.max_num_seqs, dtype=np.intp)
        self.graph_runner = None
        if self.device.type == "cuda" and not cfg.enforce_eager:
            from helix_amd.engine.graph_runner import CUDAGraphRunner
            self.graph_runner = CUDAGraphRunner(
                self.model, self.kv.caches, self.decode_ws,
                cfg.max_model_len, cfg.block_size, cfg.max_num_seqs,
                self.device)

    # ------------------------------------------------------------------
    def _auto_kv_blocks(self, nkv: int) -> int:
        if self.device.type != "cuda":
            return 512
        free, total = torch.cuda.mem_get_info(self.device)
        budget = int(total * self.cfg.gpu_memory_utilization
                     - (total - free))
        budget = max(budget, 1 << 28)
        n = KVCache.blocks_for_bytes(budget, self.model_cfg.num_layers, nkv,
                                     self.model_cfg.head_dim,
                                     self.cfg.block_size)
        # cap: max_num_seqs sequences of max_model_len
        cap = (self.cfg.max_num_seqs *
               (self.cfg.max_model_len // self.cfg.block_size + 1))
        return max(1, min(n, cap))

    # ------------------------------------------------------------------
    def add_request(self, seq_id: str, prompt_ids: List[int],
                    params: SamplingParams,
                    on_token: Optional[Callable] = None) -> Sequence:
        if len(prompt_ids) >= self.cfg.max_model_len:
            raise ValueError(
                f"prompt length {len(prompt_ids)} >= max_model_len "
                f"{self.cfg.max_model_len}")
        seq = Sequence(seq_id=seq_id, prompt_ids=list(prompt_ids),
                       params=params, on_token=on_token)
        self.seqs[seq_id] = seq
        self.waiting.append(seq)
        return seq

    def cancel(self, seq_id: str):
        seq = self.seqs.get(seq_id)
        if seq is None:
            return
        if seq.status == SeqStatus.WAITING:
            self.waiting.remove(seq)
            if seq.block_table:            # mid-chunked-prefill
                self.kv.allocator.free(seq.block_table)
                seq.block_table = []
                self._release_row(seq)
        elif seq.status == SeqStatus.RUNNING:
            self.running.remove(seq)
            self.kv.allocator.free(seq.block_table)
            seq.block_table = []
            self._release_row(seq)
        seq.status = SeqStatus.CANCELLED
        seq.finish_reason = "cancelled"

    @property
    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    def num_free_blocks(self) -> int:
        return self.kv.allocator.free_count

    # ------------------------------------------------------------------
    def step(self) -> List[tuple]:
        """Run one engine iteration. Returns [(seq, token_id, finished)]."""
        prefill_batch = self._schedule_prefill()
        if prefill_batch:
            out = self._run_prefill(prefill_batch)
        elif self.running:
            out = self._run_decode()
        else:
            return []
        self.steps += 1
        events = []
        for seq, tok in out:
            finished = self._append_token(seq, tok)
            events.append((seq, tok, finished))
            if seq.on_token is not None:
                seq.on_token(seq, tok, finished)
        return events

    # ------------------------------------------------------------------
    def _schedule_prefill(self) -> List[Sequence]:
        """Admit waiting sequences for prefill. A prompt whose remaining
        tokens exceed max_prefill_tokens is processed in CHUNKS across
        steps: each chunk writes its KV and the next chunk re-enters as
        a cached-prefix suffix (query-offset kernel + paged gather), so
        one long prompt never monopolizes a step or blows activation
        memory."""
        batch: List[Sequence] = []
        tokens = 0
        bs = self.cfg.block_size
        alloc = self.kv.allocator
        self._chunk_target: Dict[str, int] = {}
        while self.waiting and len(self.running) + len(batch) < self.cfg.max_num_seqs:
            seq = self.waiting[0]
            L = len(seq.prompt_ids)
            first = not seq.block_table
            matched: List[int] = []
            if first:
                # prefix-cache match (chain hashes over full blocks)
                if self.cfg.enable_prefix_caching:
                    if not seq.block_hashes:
                        h = b""
                        for i in range(L // bs):
                            h = chain_hash(h,
                                           seq.prompt_ids[i * bs:(i + 1) * bs])
                            seq.block_hashes.append(h)
                    for h in seq.block_hashes:
                        b = alloc.lookup_hash(h)
                        if b is None:
                            break
                        matched.append(b)
                    # always leave at least one token to prefill
                    while matched and len(matched) * bs > L - 1:
                        matched.pop()
                P = len(matched) * bs
            else:
                P = seq.cached_prefix          # continuation chunk
            budget = self.cfg.max_prefill_tokens - tokens
            if budget <= 0:
                break
            T = min(L, P + budget)
            if T <= P:
                break
            if batch and T < L:
                break                           # partial chunks go alone-ish
            nblocks = (T + bs - 1) // bs
            new_blocks = nblocks - max(len(seq.block_table), len(matched))
            # +1 headroom block so the first decode step can't OOM-deadlock
            head = 1 if T == L else 0
            if not alloc.can_allocate(new_blocks + head):
                break
            if first:
                seq.block_table = [alloc.share(b) for b in matched] + \
                    alloc.allocate(new_blocks)
            else:
                seq.block_table.extend(alloc.allocate(new_blocks))
            self._assign_row(seq)
            seq.cached_prefix = P
            self._chunk_target[seq.seq_id] = T
            if T == L:
                batch.append(self.waiting.pop(0))
            else:
                batch.append(seq)               # stays queued for next chunk
            tokens += T - P
            if T < L:
                break
        return batch

    def _slot(self, seq: Sequence, pos: int) -> int:
        bs = self.cfg.block_size
        return seq.block_table[pos // bs] * bs + pos % bs

    def _assign_row(self, seq: Sequence):
        if seq.row < 0:
            seq.row = self._free_rows.pop()
        n = len(seq.block_table)
        self._bt_np[seq.row, :n] = seq.block_table
        self._bt_np[seq.row, n:] = 0

    def _release_row(self, seq: Sequence):
        if seq.row >= 0:
            self._free_rows.append(seq.row)
            seq.row = -1
        seq.pen_init = False

    def _run_prefill(self, batch: List[Sequence]):
        bs = self.cfg.block_size
        input_ids, positions, slots, cu_q, cu_k = [], [], [], [0], [0]
        gather_blk, gather_off = [], []
        targets = getattr(self, "_chunk_target", {})
        any_cached = any(s.cached_prefix for s in batch)
        for seq in batch:
            L = targets.get(seq.seq_id, len(seq.prompt_ids))
            P = seq.cached_prefix
            input_ids.extend(seq.prompt_ids[P:L])
            positions.extend(range(P, L))
            slots.extend(self._slot(seq, p) for p in range(P, L))
            cu_q.append(cu_q[-1] + L - P)
            cu_k.append(cu_k[-1] + L)
            if any_cached:
                for p in range(L):
                    gather_blk.append(seq.block_table[p // bs])
                    gather_off.append(p % bs)
        dev = self.device
        meta = PrefillMeta(
            cu_seqlens=torch.tensor(cu_q, dtype=torch.int32, device=dev),
            max_seqlen=max(targets.get(s.seq_id, len(s.prompt_ids))
                           - s.cached_prefix for s in batch),
            slot_mapping=torch.tensor(slots, dtype=torch.int64, device=dev),
            positions=torch.tensor(positions, dtype=torch.int64, device=dev),
            cu_seqlens_k=(torch.tensor(cu_k, dtype=torch.int32, device=dev)
                          if any_cached else None),
            gather_blk=(torch.tensor(gather_blk, dtype=torch.int64,
                                     device=dev) if any_cached else None),
            gather_off=(torch.tensor(gather_off, dtype=torch.int64,
                                     device=dev) if any_cached else None))
        ids = torch.tensor(input_ids, dtype=torch.int64, device=dev)
        hidden = self.model(ids, self.kv.caches, meta)
        # register this batch's newly COMPLETED full blocks in the
        # prefix cache (partial chunks register only what they wrote)
        if self.cfg.enable_prefix_caching:
            for seq in batch:
                T = targets.get(seq.seq_id, len(seq.prompt_ids))
                start = seq.cached_prefix // bs
                end = min(len(seq.block_hashes), T // bs)
                for i in range(start, end):
                    self.kv.allocator.register_hash(seq.block_table[i],
                                                    seq.block_hashes[i])
        last_rows = torch.tensor([c - 1 for c in cu_q[1:]], dtype=torch.int64,
                                 device=dev)
        logits = self.model.compute_logits(hidden[last_rows])
        tokens = self._sample(batch, logits)
        now = time.monotonic()
        out = []
        for seq, tok in zip(batch, tokens):
            T = targets.get(seq.seq_id, len(seq.prompt_ids))
            if T < len(seq.prompt_ids):
                seq.cached_prefix = T      # partial chunk: KV written,
                continue                   # stays WAITING for next chunk
            seq.status = SeqStatus.RUNNING
            seq.first_token_time = now
            self.running.append(seq)
            out.append((seq, tok))
        return out

    def _preempt_for_blocks(self):
        """Free KV by preempting the most recent running sequence back to
        waiting (recompute-on-readmit, vLLM-style). Returns the victim or
        None."""
        for w in self.waiting:             # drop mid-chunk prefill first
            if w.block_table:
                self.kv.allocator.free(w.block_table)
                w.block_table = []
                w.cached_prefix = 0
                self._release_row(w)
                return w
        if len(self.running) <= 1:
            return None
        victim = self.running.pop()          # newest first
        self.kv.allocator.free(victim.block_table)
        victim.block_table = []
        self._release_row(victim)
        # re-admit with its generated tokens folded into the prompt so the
        # next prefill recomputes the full context
        victim.prompt_ids = victim.prompt_ids + victim.output_ids
        victim.output_ids = []
        victim.cached_prefix = 0
        victim.block_hashes = []
        victim.status = SeqStatus.WAITING
        self.waiting.insert(0, victim)
        return victim

    def _run_decode(self):
        bs = self.cfg.block_size
        dev = self.device
        batch = []
        preempted = set()                     # ids preempted this step
        nb_ = 0
        ids_np, pos_np = self._ids_np, self._pos_np
        slots_np, lens_np, rows_np = self._slots_np, self._lens_np, self._rows_np
        for seq in list(self.running):
            if seq.seq_id in preempted:
                continue
            pos = seq.total_len - 1          # position of the new input token
            nblk = pos // bs + 1
            while len(seq.block_table) < nblk:
                if not self.kv.allocator.can_allocate(1):
                    victim = self._preempt_for_blocks()
                    if victim is None:
                        raise RuntimeError(
                            "KV cache exhausted and nothing to preempt")
                    preempted.add(victim.seq_id)
                    if victim is seq:
                        break                 # we preempted ourselves
                    continue
                seq.block_table.extend(self.kv.allocator.allocate(1))
                self._bt_np[seq.row, len(seq.block_table) - 1] = \
                    seq.block_table[-1]
            if seq.seq_id in preempted:
                continue
            batch.append(seq)
            ids_np[nb_] = (seq.output_ids[-1] if seq.output_ids
                           else seq.prompt_ids[-1])
            pos_np[nb_] = pos
            slots_np[nb_] = self._slot(seq, pos)
            lens_np[nb_] = pos + 1
            rows_np[nb_] = seq.row
            nb_ += 1
        if not batch:
            return []
        B = nb_
        max_blocks = max(len(s.block_table) for s in batch)
        # one C-speed gather of the persistent rows (no per-row tensors)
        bt_np = self._bt_np[rows_np[:B], :max_blocks]
        max_len = int(lens_np[:B].max())
        if self.graph_runner is not None:
            logits = self.graph_runner.run(ids_np[:B], pos_np[:B],
                                           slots_np[:B], bt_np,
                                           lens_np[:B], max_len)
        else:
            meta = DecodeMeta(
                block_tables=torch.from_numpy(bt_np).to(dev),
                seq_lens=torch.from_numpy(lens_np[:B].copy()).to(dev),
                slot_mapping=torch.from_numpy(slots_np[:B].copy()).to(dev),
                positions=torch.from_numpy(pos_np[:B].copy()).to(dev),
                max_len=max_len,
                workspace=self.decode_ws)
            ids = torch.from_numpy(ids_np[:B].copy()).to(dev)
            hidden = self.model(ids, self.kv.caches, meta)
            logits = self.model.compute_logits(hidden)
        tokens = self._sample(batch, logits)
        return list(zip(batch, tokens))

    # ------------------------------------------------------------------
    def _ensure_pen_row(self, seq: Sequence):
        """Lazy per-row penalty tables on GPU: seen-prompt bitmap +
        output-token counts, maintained incrementally so the sampling
        kernel applies penalties without any host-side set()/Counter."""
        if self._pen_counts is None:
            V = self.model_cfg.vocab_size
            R = self.cfg.max_num_seqs
            self._pen_counts = torch.zeros(R, V, dtype=torch.int32,
                                           device=self.device)
            self._pen_seen = torch.zeros(R, V, dtype=torch.uint8,
                                         device=self.device)
        if not seq.pen_init and seq.row >= 0:
            row = seq.row
            self._pen_counts[row].zero_()
            self._pen_seen[row].zero_()
            pid = torch.tensor(seq.prompt_ids, dtype=torch.int64,
                               device=self.device)
            self._pen_seen[row][pid] = 1
            if seq.output_ids:   # preemption recompute: rebuild counts
                oid = torch.tensor(seq.output_ids, dtype=torch.int64,
                                   device=self.device)
                self._pen_counts[row].index_put_(
                    (oid,), torch.ones(len(seq.output_ids),
                                       dtype=torch.int32,
                                       device=self.device),
                    accumulate=True)
            seq.pen_init = True

    def _apply_json_masks(self, batch: List[Sequence],
                          logits: torch.Tensor):
        """Grammar-constrained decoding (response_format json_object):
        mask each JSON row's logits to the automaton's allowed byte
        tokens; a complete document admits only EOS."""
        from helix_amd.engine.json_mode import JSONByteMask
        from helix_amd.utils.tokenizer import N_SPECIAL
        for i, s in enumerate(batch):
            if not s.params.json_mode:
                continue
            if s.json_mask is None:
                s.json_mask = JSONByteMask()
            m = s.json_mask
            if m.complete:
                allowed = [self.cfg.eos_token_id]                     if self.cfg.eos_token_id >= 0 else []
            else:
                V = logits.shape[1]
                allowed = [b + N_SPECIAL for b in m.allowed_bytes()
                           if b + N_SPECIAL < V]
            row = logits[i]
            if not allowed:
                continue
            idx = torch.tensor(allowed, dtype=torch.int64,
                               device=row.device)
            keep = row[idx].clone()
            row.fill_(float("-inf"))
            row[idx] = keep

    def _advance_json_masks(self, batch: List[Sequence],
                            out: List[int]):
        from helix_amd.utils.tokenizer import N_SPECIAL
        for i, s in enumerate(batch):
            if s.params.json_mode and s.json_mask is not None:
                b = out[i] - N_SPECIAL
                if 0 <= b < 256:
                    s.json_mask.push_byte(b)

    def _sample(self, batch: List[Sequence], logits: torch.Tensor) -> List[int]:
        # wait: decode path's new input token is appended by _append_token;
        # here logits are [B, V].
        B = len(batch)
        needs_bias = any(s.params.logit_bias for s in batch)
        needs_json = any(s.params.json_mode for s in batch)
        if needs_bias or needs_json:
            # logits may be an inference-mode tensor (graph/forward
            # output): clone before in-place bias / grammar masking
            logits = logits.clone()
        if needs_bias:
            for i, s in enumerate(batch):
                lb = s.params.logit_bias
                if not lb:
                    continue
                idx = torch.tensor([int(k) for k in lb],
                                   dtype=torch.int64,
                                   device=logits.device)
                vals = torch.tensor([float(v) for v in lb.values()],
                                    dtype=logits.dtype,
                                    device=logits.device)
                logits[i].index_add_(0, idx, vals)
        if needs_json:
            self._apply_json_masks(batch, logits)
        lp_rows = [i for i, s_ in enumerate(batch) if s_.params.logprobs]
        needs_proc = any(s.params.needs_logit_processing for s in batch)
        gpu_fast = logits.is_cuda and needs_proc
        if needs_proc and not gpu_fast:
            # float() is a no-op on fp32 CPU engines, and the model
            # forward runs under inference_mode: clone so the in-place
            # top-k/top-p/penalty edits are legal
            logits = self._process_logits(
                batch, logits.float().clone()
                if logits.is_inference() else logits.float())
        elif gpu_fast and lp_rows:
            # logprob rows report over the FILTERED distribution, so they
            # keep the reference host path; their kernel params are
            # neutralized below (no double filtering/penalties)
            sub = logits[lp_rows].float().clone()
            self._process_logits([batch[i] for i in lp_rows], sub)
            if logits.is_inference():
                logits = logits.clone()
            logits[lp_rows] = sub.to(logits.dtype)
        on_gpu = logits.is_cuda
        if on_gpu:
            hf = self._samp_hf.numpy()
            hs = self._samp_hs.numpy()
        else:
            hf = np.empty((5, B), dtype=np.float32)
            hs = np.empty(B, dtype=np.int64)
        # The decode batch composition is stable between steps (every
        # running row appends exactly one token); the per-row parameter
        # staging loops are ~2x O(B) Python per step — cache them keyed
        # by (seq ids, lp rows) and just advance the seed vector.
        sig = (tuple(s.seq_id for s in batch), tuple(lp_rows),
               gpu_fast, on_gpu)
        cache = getattr(self, "_samp_cache", None)
        if cache is not None and cache["sig"] == sig:
            cache["outlens"] += 1
            hs[:B] = (cache["base"] + cache["outlens"] * _SEED_MIX)                 & 0x7FFFFFFFFFFFFFFF
            hf[0, :B] = cache["temps"]
        else:
            base_v = np.empty(B, dtype=np.uint64)
            outlen_v = np.empty(B, dtype=np.uint64)
            for i, s in enumerate(batch):
                hf[0, i] = s.params.temperature
                # Fallback seed must be deterministic across processes:
                # SPMD TP ranks each run this code and must draw
                # identical Gumbel noise. hash() is per-process
                # randomized (PYTHONHASHSEED) — use crc32.
                base_v[i] = s.params.seed if s.params.seed is not None                     else (zlib.crc32(s.seq_id.encode()) & 0x7FFFFFFF)
                outlen_v[i] = len(s.output_ids)
            hs[:B] = (base_v + outlen_v * _SEED_MIX)                 & 0x7FFFFFFFFFFFFFFF
            self._samp_cache = {"sig": sig, "base": base_v,
                                "outlens": outlen_v,
                                "temps": hf[0, :B].copy(),
                                "params_done": False}
            cache = self._samp_cache
        if gpu_fast:
            hi = self._samp_hi.numpy()
            if cache.get("params_done"):
                pen_idx = cache["pen_idx"]
                hf[1:5, :B] = cache["pf"]
                hi[:, :B] = cache["pi"]
            else:
                lp_set = set(lp_rows)
                hf[1, :B] = 1.0   # top_p
                hf[2, :B] = 1.0   # repetition
                hf[3, :B] = 0.0   # presence
                hf[4, :B] = 0.0   # frequency
                hi[0, :B] = 0     # top_k
                hi[1, :B] = -1    # row_map
                pen_idx = []
                for i, s in enumerate(batch):
                    if i in lp_set:
                        continue            # host-processed already
                    p = s.params
                    hf[1, i] = p.top_p
                    hi[0, i] = p.top_k
                    if (p.repetition_penalty != 1.0
                            or p.presence_penalty != 0.0
                            or p.frequency_penalty != 0.0):
                        hf[2, i] = p.repetition_penalty
                        hf[3, i] = p.presence_penalty
                        hf[4, i] = p.frequency_penalty
                        self._ensure_pen_row(s)
                        hi[1, i] = s.row
                        pen_idx.append(i)
                cache["pen_idx"] = pen_idx
                cache["pf"] = hf[1:5, :B].copy()
                cache["pi"] = hi[:, :B].copy()
                cache["params_done"] = True
            dev = logits.device
            self._samp_df[:, :B].copy_(self._samp_hf[:, :B],
                                       non_blocking=True)
            self._samp_ds[:B].copy_(self._samp_hs[:B], non_blocking=True)
            self._samp_di[:, :B].copy_(self._samp_hi[:, :B],
                                       non_blocking=True)
            toks = ops.sample_tokens_ext(
                logits.contiguous(), self._samp_df[0, :B],
                self._samp_ds[:B], self._samp_df[1, :B],
                self._samp_di[0, :B], self._samp_df[2, :B],
                self._samp_df[3, :B], self._samp_df[4, :B],
                self._pen_counts, self._pen_seen, self._samp_di[1, :B])
            if pen_idx:
                # incremental count update with the just-sampled tokens
                rows_t = torch.tensor([batch[i].row for i in pen_idx],
                                      dtype=torch.int64, device=dev)
                sel = toks[pen_idx]
                self._pen_counts.index_put_(
                    (rows_t, sel), torch.ones(len(pen_idx),
                                              dtype=torch.int32, device=dev),
                    accumulate=True)
        elif on_gpu:
            self._samp_df[0, :B].copy_(self._samp_hf[0, :B],
                                       non_blocking=True)
            self._samp_ds[:B].copy_(self._samp_hs[:B], non_blocking=True)
            toks = ops.sample_tokens(logits.contiguous(),
                                     self._samp_df[0, :B],
                                     self._samp_ds[:B])
        else:
            temps = torch.from_numpy(hf[0, :B].copy())
            seeds_t = torch.from_numpy(hs[:B].copy())
            toks = ops.sample_tokens(logits.contiguous(), temps, seeds_t)
        out = toks.cpu().tolist()
        if any(s.params.json_mode for s in batch):
            self._advance_json_masks(batch, out)
        # top-k logprobs for sequences that requested them (one extra
        # log_softmax + topk over just those rows)
        lp_rows = [i for i, s_ in enumerate(batch) if s_.params.logprobs]
        if lp_rows:
            k = max(batch[i].params.logprobs for i in lp_rows)
            sub = logits[lp_rows].float()
            lsm = torch.log_softmax(sub, dim=-1)
            topv, topi = lsm.topk(min(k, lsm.shape[-1]), dim=-1)
            topv = topv.cpu().tolist()
            topi = topi.cpu().tolist()
            for j, i in enumerate(lp_rows):
                tok = out[i]
                tok_lp = float(lsm[j, tok])
                kk = batch[i].params.logprobs
                batch[i].logprobs.append({
                    "token": tok, "logprob": tok_lp,
                    "top_logprobs": [
                        {"token": t, "logprob": v}
                        for t, v in zip(topi[j][:kk], topv[j][:kk])]})
        return out

    def _process_logits(self, batch, logits: torch.Tensor) -> torch.Tensor:
        for i, seq in enumerate(batch):
            p = seq.params
            if p.repetition_penalty != 1.0 or p.presence_penalty != 0.0 \
               or p.frequency_penalty != 0.0:
                seen = torch.tensor(
                    list(set(seq.prompt_ids + seq.output_ids)),
                    dtype=torch.int64, device=logits.device)
                row = logits[i]
                if p.repetition_penalty != 1.0:
                    vals = row[seen]
                    row[seen] = torch.where(vals > 0,
                                            vals / p.repetition_penalty,
                                            vals * p.repetition_penalty)
                if p.presence_penalty != 0.0:
                    row[seen] -= p.presence_penalty
                if p.frequency_penalty != 0.0:
                    from collections import Counter
                    cnt = Counter(seq.output_ids)
                    idx = torch.tensor(list(cnt.keys()), dtype=torch.int64,
                                       device=logits.device)
                    freq = torch.tensor(list(cnt.values()),
                                        dtype=torch.float32,
                                        device=logits.device)
                    row[idx] -= p.frequency_penalty * freq
            if p.top_k > 0:
                kth = torch.topk(logits[i], p.top_k).values[-1]
                logits[i][logits[i] < kth] = -float("inf")
            if p.top_p < 1.0:
                sorted_logits, idx = torch.sort(logits[i], descending=True)
                probs = torch.softmax(sorted_logits, -1)
                cum = probs.cumsum(-1)
                cut = (cum - probs) >= p.top_p   # keep first token past p
                sorted_logits[cut] = -float("inf")
                logits[i].scatter_(0, idx, sorted_logits)
        return logits

    def _append_token(self, seq: Sequence, tok: int) -> bool:
        seq.output_ids.append(tok)
        p = seq.params
        finished = None
        if not p.ignore_eos and (tok == self.cfg.eos_token_id
                                 or tok in p.stop_token_ids):
            finished = "stop"
        elif len(seq.output_ids) >= p.max_tokens:
            finished = "length"
        elif seq.total_len >= self.cfg.max_model_len:
            finished = "length"
        if finished:
            seq.status = SeqStatus.FINISHED
            seq.finish_reason = finished
            self.running.remove(seq)
            self.kv.allocator.free(seq.block_table)
            seq.block_table = []
            self._release_row(seq)
            return True
        return False

    # ------------------------------------------------------------------
    def generate(self, prompts: List[List[int]],
                 params: SamplingParams) -> List[List[int]]:
        """Synchronous batch generate (used by tests and bench)."""
        # ids must be identical across SPMD TP ranks: the seedless
        # sampling fallback seeds from crc32(seq_id), so an id built
        # from a memory address (id(prompts)) silently diverges ranks
        self._gen_calls = getattr(self, "_gen_calls", 0) + 1
        ids = [f"gen-{self._gen_calls}-{i}"
               for i in range(len(prompts))]
        for sid, p in zip(ids, prompts):
            self.add_request(sid, p, params)
        while self.has_work:
            self.step()
        return [self.seqs[sid].output_ids for sid in ids]
