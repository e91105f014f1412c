"""Tensor-parallel correctness over gloo (world_size=2, CPU).

TP-sharded forward must match the full single-process model (the same
invariant the GPU RCCL path relies on).
"""
import os

import socket

import pytest
import torch
import torch.multiprocessing as mp


def _free_port() -> str:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return str(p)


def _spawn2(target, mkargs, timeout=300, attempts=2):
    """Run a 2-rank spawn job; retry once on nonzero exits (rare gloo
    rendezvous flakes under load). mkargs(rank, port) -> args tuple."""
    ctx = mp.get_context("spawn")
    for attempt in range(attempts):
        port = _free_port()
        procs = [ctx.Process(target=target, args=mkargs(r, port))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=timeout)
        codes = [p.exitcode for p in procs]
        for p in procs:
            if p.is_alive():
                p.terminate()
        if all(c == 0 for c in codes):
            return
        if attempt == attempts - 1:
            raise AssertionError(f"spawn ranks failed: {codes}")

from helix_amd.models.llama import PRESETS, LlamaForCausalLM, PrefillMeta
from helix_amd.parallel import shard_llama_state_dict


def _full_forward(cfg_name, ids, positions, cu, max_len):
    cfg = PRESETS[cfg_name]
    torch.manual_seed(0)
    model = LlamaForCausalLM(cfg).float()
    model.init_random(0)
    meta = PrefillMeta(cu_seqlens=cu, max_seqlen=max_len,
                       slot_mapping=torch.full((ids.shape[0],), -1,
                                               dtype=torch.int64),
                       positions=positions)
    hidden = model(ids, None, meta)
    return model.compute_logits(hidden), model.state_dict()


def _tp_worker(rank, world, cfg_name, ids, positions, cu, max_len, sd,
               out_path, port):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=port,
                      LOCAL_RANK=str(rank))
    from helix_amd import parallel
    parallel.init_tp(world, backend="gloo")
    cfg = PRESETS[cfg_name]
    model = LlamaForCausalLM(cfg, tp_size=world, tp_rank=rank).float()
    model.load_state_dict(shard_llama_state_dict(sd, cfg, world, rank),
                          strict=True)
    meta = PrefillMeta(cu_seqlens=cu, max_seqlen=max_len,
                       slot_mapping=torch.full((ids.shape[0],), -1,
                                               dtype=torch.int64),
                       positions=positions)
    hidden = model(ids, None, meta)
    logits = model.compute_logits(hidden)
    if rank == 0 and out_path:
        torch.save(logits.detach(), out_path)
    torch.distributed.destroy_process_group()


@pytest.mark.parametrize("cfg_name", ["tiny-gqa"])
def test_tp2_matches_full(cfg_name):
    torch.manual_seed(42)
    lens = [9, 5]
    T = sum(lens)
    cfg = PRESETS[cfg_name]
    ids = torch.randint(0, cfg.vocab_size, (T,))
    positions = torch.cat([torch.arange(l) for l in lens])
    cu = torch.tensor([0, lens[0], T], dtype=torch.int32)

    full_logits, sd = _full_forward(cfg_name, ids, positions, cu, max(lens))

    import tempfile
    with tempfile.TemporaryDirectory() as td:
        out_path = os.path.join(td, "logits.pt")
        _spawn2(_tp_worker,
                lambda r, port: (r, 2, cfg_name, ids, positions, cu,
                                 max(lens), sd, out_path, port))
        tp_logits = torch.load(out_path)
    torch.testing.assert_close(tp_logits, full_logits, atol=2e-3, rtol=2e-3)


def test_shard_shapes():
    cfg = PRESETS["tiny-gqa"]
    model = LlamaForCausalLM(cfg)
    sd = model.state_dict()
    for r in range(2):
        shard = shard_llama_state_dict(sd, cfg, 2, r)
        m2 = LlamaForCausalLM(cfg, tp_size=2, tp_rank=r)
        m2.load_state_dict(shard, strict=True)


def _tp_engine_worker(rank, world, sd, prompts, out_path, port):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=port,
                      LOCAL_RANK=str(rank))
    from helix_amd import parallel
    from helix_amd.engine.engine import EngineConfig, LLMEngine
    from helix_amd.engine.sampling_params import SamplingParams
    parallel.init_tp(world, backend="gloo")
    cfg = PRESETS["tiny-gqa"]
    model = LlamaForCausalLM(cfg, tp_size=world, tp_rank=rank).float()
    model.load_state_dict(shard_llama_state_dict(sd, cfg, world, rank),
                          strict=True)
    eng = LLMEngine(EngineConfig(model="tiny-gqa", max_model_len=256,
                                 max_num_seqs=4, kv_cache_blocks=128,
                                 eos_token_id=-1),
                    device="cpu", model=model, tp_size=world, tp_rank=rank)
    out = eng.generate(prompts, SamplingParams(temperature=0.0,
                                               max_tokens=8,
                                               ignore_eos=True))
    if rank == 0 and out_path:
        torch.save(out, out_path)
    torch.distributed.destroy_process_group()


def test_tp2_engine_generate_matches_single():
    """Full SPMD-TP engine loop over gloo: the sharded 2-rank engine must
    produce the exact greedy tokens of the single-process engine."""
    import tempfile
    from helix_amd.engine.engine import EngineConfig, LLMEngine
    from helix_amd.engine.sampling_params import SamplingParams
    cfg = PRESETS["tiny-gqa"]
    torch.manual_seed(0)
    full = LlamaForCausalLM(cfg).float()
    full.init_random(0)
    sd = full.state_dict()
    prompts = [[1, 2, 3, 4, 5], [9, 8, 7]]
    eng = LLMEngine(EngineConfig(model="tiny-gqa", max_model_len=256,
                                 max_num_seqs=4, kv_cache_blocks=128,
                                 eos_token_id=-1),
                    device="cpu", model=full)
    want = eng.generate(prompts, SamplingParams(temperature=0.0,
                                                max_tokens=8,
                                                ignore_eos=True))
    with tempfile.TemporaryDirectory() as td:
        out_path = os.path.join(td, "out.pt")
        _spawn2(_tp_engine_worker,
                lambda r, port: (r, 2, sd, prompts, out_path, port))
        got = torch.load(out_path)
    assert got == want


def _tp_temp_worker(rank, world, sd, prompts, out_path, port):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=port,
                      LOCAL_RANK=str(rank))
    from helix_amd import parallel
    from helix_amd.engine.engine import EngineConfig, LLMEngine
    from helix_amd.engine.sampling_params import SamplingParams
    parallel.init_tp(world, backend="gloo")
    cfg = PRESETS["tiny-gqa"]
    model = LlamaForCausalLM(cfg, tp_size=world, tp_rank=rank).float()
    model.load_state_dict(shard_llama_state_dict(sd, cfg, world, rank),
                          strict=True)
    eng = LLMEngine(EngineConfig(model="tiny-gqa", max_model_len=256,
                                 max_num_seqs=4, kv_cache_blocks=128,
                                 eos_token_id=-1),
                    device="cpu", model=model, tp_size=world, tp_rank=rank)
    out = eng.generate(prompts, SamplingParams(temperature=0.8,
                                               max_tokens=8, seed=77,
                                               ignore_eos=True))
    # EVERY rank writes: both must have sampled identical tokens
    torch.save(out, f"{out_path}.{rank}")
    torch.distributed.destroy_process_group()


def test_tp2_temperature_sampling_lockstep():
    """At temperature>0, both TP ranks must sample the same tokens (the
    seed-driven Gumbel sampler is the lockstep mechanism) and match the
    single-process engine."""
    import tempfile

    from helix_amd.engine.engine import EngineConfig, LLMEngine
    from helix_amd.engine.sampling_params import SamplingParams
    cfg = PRESETS["tiny-gqa"]
    torch.manual_seed(0)
    full = LlamaForCausalLM(cfg).float()
    full.init_random(0)
    sd = full.state_dict()
    prompts = [[1, 2, 3, 4, 5], [9, 8, 7]]
    eng = LLMEngine(EngineConfig(model="tiny-gqa", max_model_len=256,
                                 max_num_seqs=4, kv_cache_blocks=128,
                                 eos_token_id=-1),
                    device="cpu", model=full)
    want = eng.generate(prompts, SamplingParams(temperature=0.8,
                                                max_tokens=8, seed=77,
                                                ignore_eos=True))
    with tempfile.TemporaryDirectory() as td:
        out_path = os.path.join(td, "out.pt")
        _spawn2(_tp_temp_worker,
                lambda r, port: (r, 2, sd, prompts, out_path, port))
        r0 = torch.load(f"{out_path}.0")
        r1 = torch.load(f"{out_path}.1")
    assert r0 == r1 == want


def _tp_feature_worker(rank, world, sd, prompts, out_path, port):
    """Rank worker exercising the round-2 sampling features under
    SPMD: seedless temperature sampling over MULTIPLE steps (the
    cached staging must advance seeds identically on every rank),
    JSON-mode masking, and logit_bias."""
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=port,
                      LOCAL_RANK=str(rank))
    from helix_amd import parallel
    from helix_amd.engine.engine import EngineConfig, LLMEngine
    from helix_amd.engine.sampling_params import SamplingParams
    parallel.init_tp(world, backend="gloo")
    cfg = PRESETS["tiny-gqa"]
    model = LlamaForCausalLM(cfg, tp_size=world, tp_rank=rank).float()
    model.load_state_dict(shard_llama_state_dict(sd, cfg, world, rank),
                          strict=True)
    eng = LLMEngine(EngineConfig(model="tiny-gqa", max_model_len=256,
                                 max_num_seqs=4, kv_cache_blocks=128,
                                 eos_token_id=-1),
                    device="cpu", model=model, tp_size=world,
                    tp_rank=rank)
    outs = []
    # seedless sampling (deterministic crc32 fallback + cached
    # staging advancing outlens per step)
    outs.append(eng.generate(
        prompts, SamplingParams(temperature=0.9, top_p=0.8,
                                max_tokens=10, ignore_eos=True)))
    # json grammar masking
    outs.append(eng.generate(
        [prompts[0]], SamplingParams(temperature=1.0, seed=None,
                                     max_tokens=10, ignore_eos=True,
                                     json_mode=True)))
    # logit bias pins the output
    outs.append(eng.generate(
        [prompts[1]], SamplingParams(temperature=0.7, max_tokens=4,
                                     ignore_eos=True,
                                     logit_bias={33: 500.0})))
    if out_path:
        torch.save(outs, out_path + f".r{rank}")
    torch.distributed.destroy_process_group()


def test_tp2_round2_sampling_features_lockstep():
    """Both ranks must emit IDENTICAL tokens for seedless sampling,
    JSON mode, and logit_bias — any divergence silently corrupts TP
    state (ADVICE r1 high-severity class of bug)."""
    import tempfile
    cfg = PRESETS["tiny-gqa"]
    torch.manual_seed(0)
    full = LlamaForCausalLM(cfg).float()
    full.init_random(0)
    sd = full.state_dict()
    prompts = [[1, 2, 3, 4, 5], [9, 8, 7]]
    with tempfile.TemporaryDirectory() as td:
        out_path = os.path.join(td, "out.pt")
        _spawn2(_tp_feature_worker,
                lambda r, port: (r, 2, sd, prompts, out_path, port))
        r0 = torch.load(out_path + ".r0")
        r1 = torch.load(out_path + ".r1")
    assert r0 == r1, "rank outputs diverged"
    assert r0[2][0] == [33, 33, 33, 33]    # bias honored under TP
