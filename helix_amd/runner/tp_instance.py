"""Tensor-parallel serving instance: the runner-side daemon that serves
one TP-sharded model from N spawned per-GPU worker processes.

The reference delegates multi-GPU serving to vLLM's internal executor
(SURVEY.md §2.8 "Model containers"); here it is a native runner
component: one process per GPU (RCCL over xGMI; gloo on CPU for tests),
SPMD engine loop on every rank, commands fanned out by rank 0 via
``broadcast_object_list`` so all ranks execute the identical
add_request/cancel/step sequence, and token events flowing back to the
serving process over a multiprocessing queue.

The public surface (``submit`` / ``cancel`` / ``in_flight`` /
``shutdown``) matches ``runner.service.LLMInstance`` so the OpenAI
adapter and RunnerService treat TP and single-GPU instances uniformly.
"""
from __future__ import annotations

import logging
import os
import queue as _queue
import socket
import threading
import time
from typing import Dict, List, Optional

import torch

from helix_amd.engine.sampling_params import SamplingParams

log = logging.getLogger("helix_amd.tp_instance")


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


class _SeqProxy:
    """What main-process on_token callbacks see (seq_id + finish_reason,
    the two fields adapters read)."""
    __slots__ = ("seq_id", "finish_reason")

    def __init__(self, seq_id: str, finish_reason: Optional[str]):
        self.seq_id = seq_id
        self.finish_reason = finish_reason


def _tp_worker(rank: int, world: int, preset: str, engine_kwargs: dict,
               master_port: int, backend: str, device_type: str,
               sd_path: Optional[str], seed: int, cmd_q, evt_q):
    """One TP rank: SPMD engine loop driven by rank-0 command broadcasts."""
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(master_port))
    import torch.distributed as dist
    from helix_amd import parallel
    from helix_amd.engine.engine import EngineConfig, LLMEngine
    from helix_amd.models.llama import LlamaForCausalLM, PRESETS

    try:
        if device_type == "cuda":
            torch.cuda.set_device(rank)
            device = f"cuda:{rank}"
        else:
            device = "cpu"
        parallel.init_tp(world, backend=backend)
        cfg = PRESETS[preset]
        model = None
        if sd_path is not None:
            # shard a known full checkpoint (tests / real weights)
            from helix_amd.parallel import shard_llama_state_dict
            full_sd = torch.load(sd_path, map_location="cpu")
            model = LlamaForCausalLM(cfg, tp_size=world, tp_rank=rank)
            if device_type == "cuda":
                model = model.to(torch.bfloat16)
            else:
                model = model.float()
            model.load_state_dict(
                {k: v.to(next(iter(model.state_dict().values())).dtype)
                 for k, v in
                 shard_llama_state_dict(full_sd, cfg, world, rank).items()},
                strict=True)
            model = model.to(device)
        ecfg = EngineConfig(model=preset, seed=seed, **engine_kwargs)
        eng = LLMEngine(ecfg, device=device, model=model,
                        tp_size=world, tp_rank=rank)

        def cb(seq, tok, fin):
            evt_q.put(("tok", seq.seq_id, tok, fin, seq.finish_reason))

        if rank == 0:
            evt_q.put(("ready", None, None, None, None))
        while True:
            if rank == 0:
                cmds = []
                timeout = 0.0 if eng.has_work else 0.05
                try:
                    cmds.append(cmd_q.get(timeout=timeout))
                except _queue.Empty:
                    pass
                while True:
                    try:
                        cmds.append(cmd_q.get_nowait())
                    except _queue.Empty:
                        break
                obj = [cmds]
            else:
                obj = [None]
            dist.broadcast_object_list(obj, src=0)
            stop = False
            for c in obj[0]:
                if c[0] == "stop":
                    stop = True
                elif c[0] == "submit":
                    _, seq_id, ids, params = c
                    eng.add_request(seq_id, ids, params,
                                    on_token=cb if rank == 0 else None)
                elif c[0] == "cancel":
                    existed = c[1] in eng.seqs and \
                        eng.seqs[c[1]].finish_reason is None
                    eng.cancel(c[1])
                    if rank == 0 and existed:
                        # the engine does not emit an event on cancel
                        # (single-process adapters return immediately);
                        # across processes the serving side needs closure
                        # to release its callback + inflight slot
                        evt_q.put(("tok", c[1], 0, True, "cancelled"))
            if stop:
                break
            if eng.has_work:
                eng.step()
        dist.destroy_process_group()
    except Exception as e:  # surface worker death to the serving process
        log.exception("tp worker %d died", rank)
        if rank == 0:
            try:
                evt_q.put(("dead", None, None, None, str(e)))
            except Exception:
                pass
        raise


class TPLLMInstance:
    """Drop-in LLMInstance replacement backed by `tp_size` worker
    processes (one per GPU)."""

    def __init__(self, spec, tp_size: int, device_type: Optional[str] = None,
                 backend: Optional[str] = None, sd_path: Optional[str] = None,
                 seed: int = 0, start_timeout: float = 600.0):
        self.spec = spec
        self.tp_size = tp_size
        if device_type is None:
            device_type = "cuda" if torch.cuda.is_available() else "cpu"
        if backend is None:
            backend = "nccl" if device_type == "cuda" else "gloo"
        ctx = torch.multiprocessing.get_context("spawn")
        self.cmd_q = ctx.Queue()
        self.evt_q = ctx.Queue()
        engine_kwargs = dict(
            max_model_len=spec.max_model_len,
            max_num_seqs=spec.max_num_seqs,
            kv_cache_blocks=spec.kv_cache_blocks,
            quantization=getattr(spec, "quantization", None),
            kv_cache_dtype=getattr(spec, "kv_cache_dtype", "bf16"),
        )
        if getattr(spec, "eos_token_id", None) is not None:
            engine_kwargs["eos_token_id"] = spec.eos_token_id
        port = _free_port()
        self.procs = [
            ctx.Process(target=_tp_worker,
                        args=(r, tp_size, spec.preset, engine_kwargs, port,
                              backend, device_type, sd_path, seed,
                              self.cmd_q, self.evt_q),
                        daemon=True)
            for r in range(tp_size)
        ]
        for p in self.procs:
            p.start()
        # wait for rank0's engine to come up (weights + graphs)
        deadline = time.time() + start_timeout
        while True:
            try:
                evt = self.evt_q.get(timeout=1.0)
            except _queue.Empty:
                if time.time() > deadline:
                    self.shutdown()
                    raise RuntimeError("TP workers failed to start")
                if any(p.exitcode not in (None, 0) for p in self.procs):
                    self.shutdown()
                    raise RuntimeError("TP worker died during startup")
                continue
            if evt[0] == "ready":
                break
            if evt[0] == "dead":
                self.shutdown()
                raise RuntimeError(f"TP worker failed: {evt[4]}")
        self._cbs: Dict[str, object] = {}
        self._lock = threading.Lock()
        self._inflight = 0
        self.last_used = time.time()
        self.stop = False
        self._pump = threading.Thread(target=self._pump_events, daemon=True,
                                      name=f"tp-pump-{spec.name}")
        self._pump.start()

    @property
    def in_flight(self) -> int:
        return self._inflight

    def _pump_events(self):
        while not self.stop:
            try:
                evt = self.evt_q.get(timeout=0.2)
            except _queue.Empty:
                continue
            kind = evt[0]
            if kind == "tok":
                _, seq_id, tok, fin, reason = evt
                with self._lock:
                    cb = self._cbs.get(seq_id)
                    if fin:
                        self._cbs.pop(seq_id, None)
                        self._inflight = max(0, self._inflight - 1)
                if cb is not None:
                    try:
                        cb(_SeqProxy(seq_id, reason), tok, fin)
                    except Exception:
                        log.exception("on_token callback failed")
            elif kind == "dead":
                self._on_dead(evt[4])

    def _on_dead(self, reason: str):
        """Engine fleet died: error-finish every in-flight callback."""
        log.error("TP engine died: %s", reason)
        with self._lock:
            cbs, self._cbs = self._cbs, {}
            self._inflight = 0
        for seq_id, cb in cbs.items():
            try:
                cb(_SeqProxy(seq_id, f"error: {reason}"), 0, True)
            except Exception:
                pass

    def submit(self, seq_id: str, prompt_ids: List[int],
               params: SamplingParams, on_token) -> None:
        self.last_used = time.time()
        with self._lock:
            if on_token is not None:
                self._cbs[seq_id] = on_token
            self._inflight += 1
        self.cmd_q.put(("submit", seq_id, list(prompt_ids), params))

    def cancel(self, seq_id: str):
        self.cmd_q.put(("cancel", seq_id))

    def shutdown(self):
        self.stop = True
        # a killed worker can die holding the queue's internal lock, so
        # only enqueue the stop command while the whole fleet is alive
        # (otherwise terminate directly — nothing would consume it)
        if all(p.is_alive() for p in self.procs):
            try:
                self.cmd_q.put(("stop",))
            except Exception:
                pass
            for p in self.procs:
                p.join(timeout=30)
        for p in self.procs:
            if p.is_alive():
                p.terminate()
        for p in self.procs:
            p.join(timeout=10)
        self.cmd_q.cancel_join_thread()
        self.evt_q.cancel_join_thread()
        if hasattr(self, "_pump"):
            self._pump.join(timeout=5)
