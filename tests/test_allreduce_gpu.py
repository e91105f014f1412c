"""GPU tests for the one-shot xGMI allreduce (ops/hip/allreduce.hip).

Two spawned processes SHARE cuda:0 (hipIpc maps same-device mailboxes,
which RCCL cannot do with two ranks), exchange handles over gloo, and
must produce the exact elementwise bf16 sum — eager and under hipGraph
capture/replay. This is the production TP-decode allreduce path
(SURVEY.md §2.6 one-shot plan; replaces NCCL-inside-vLLM,
reference design/2026-04-28-cloud-gpu-smoke-results.md:28).
"""
import os
import socket

import pytest
import torch

pytestmark = pytest.mark.gpu


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _ar_worker(rank: int, world: int, port: int, mode: str, out_q):
    try:
        os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                          LOCAL_RANK="0", MASTER_ADDR="127.0.0.1",
                          MASTER_PORT=str(port))
        import torch.distributed as dist
        torch.cuda.set_device(0)
        dist.init_process_group("gloo")
        from helix_amd import ops
        cap = 8 << 20
        handle = ops._native().ar_create(world, rank, cap)
        gathered = [None] * world
        dist.all_gather_object(gathered, handle.numpy().tobytes())
        ops._native().ar_open(
            [torch.frombuffer(bytearray(b), dtype=torch.uint8)
             for b in gathered])

        results = []
        sizes = [8, 128, 4096, 4096 * 64 + 3, 1 << 20]
        if mode == "eager":
            for n in sizes:
                g = torch.Generator().manual_seed(1000 + n)
                xs = [torch.randn(n, generator=g).bfloat16()
                      for _ in range(world)]
                want = sum(x.float() for x in xs).bfloat16()
                x = xs[rank].cuda()
                ops._native().ar_allreduce(x, x)
                torch.cuda.synchronize()
                results.append(bool(torch.equal(x.cpu(), want)))
        else:  # graph capture + replay with changing inputs
            n = 4096 * 16
            static = torch.zeros(n, dtype=torch.bfloat16, device="cuda")
            out = torch.zeros_like(static)
            # warmup on a side stream (same count on every rank)
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                ops._native().ar_allreduce(static, out)
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                ops._native().ar_allreduce(static, out)
            for rep in range(3):
                g = torch.Generator().manual_seed(7 + rep)
                xs = [torch.randn(n, generator=g).bfloat16()
                      for _ in range(world)]
                want = sum(x.float() for x in xs).bfloat16()
                static.copy_(xs[rank].cuda())
                graph.replay()
                torch.cuda.synchronize()
                results.append(bool(torch.equal(out.cpu(), want)))
        ops._native().ar_destroy()
        dist.destroy_process_group()
        out_q.put((rank, results))
    except Exception as e:  # pragma: no cover
        out_q.put((rank, f"error: {e}"))
        raise


def _run_2proc(mode: str):
    ctx = torch.multiprocessing.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [ctx.Process(target=_ar_worker, args=(r, 2, port, mode, q),
                         daemon=True) for r in range(2)]
    for p in procs:
        p.start()
    outs = {}
    for _ in range(2):
        rank, res = q.get(timeout=240)
        outs[rank] = res
    for p in procs:
        p.join(timeout=60)
    return outs


@pytest.mark.timeout(420)
def test_one_shot_allreduce_two_procs_one_gpu():
    outs = _run_2proc("eager")
    for rank, res in outs.items():
        assert isinstance(res, list), res
        assert all(res), (rank, res)


@pytest.mark.timeout(420)
def test_one_shot_allreduce_hipgraph_replay():
    outs = _run_2proc("graph")
    for rank, res in outs.items():
        assert isinstance(res, list), res
        assert all(res), (rank, res)


@pytest.mark.timeout(600)
def test_tp2_instance_one_gpu_matches_single_process(tmp_path):
    """Full TP=2 serving equivalence on ONE GPU: both ranks share cuda:0
    (gloo control PG + one-shot IPC allreduce for the decode path, with
    hipGraph capture enabled because the allreduce kernel is
    graph-capturable). Tokens must equal a single-process engine run of
    the unsharded model."""
    from helix_amd.engine.engine import EngineConfig, LLMEngine
    from helix_amd.engine.sampling_params import SamplingParams
    from helix_amd.models.llama import LlamaForCausalLM, PRESETS
    from helix_amd.runner.service import ModelSpec
    from helix_amd.runner.tp_instance import TPLLMInstance
    import threading

    cfg = PRESETS["tiny-gqa"]
    torch.manual_seed(0)
    full = LlamaForCausalLM(cfg).to(torch.bfloat16)
    full.init_random(0)
    sd_path = str(tmp_path / "full_sd.pt")
    torch.save(full.state_dict(), sd_path)

    prompts = [[1, 2, 3, 4, 5], [9, 8, 7]]
    sp = SamplingParams(temperature=0.0, max_tokens=8, ignore_eos=True)
    eng = LLMEngine(EngineConfig(model="tiny-gqa", max_model_len=256,
                                 max_num_seqs=4, kv_cache_blocks=128,
                                 eos_token_id=-1),
                    device="cuda:0", model=full.cuda())
    eng.add_request("req-0", prompts[0], sp)
    eng.add_request("req-1", prompts[1], sp)
    while eng.has_work:
        eng.step()
    want = [eng.seqs["req-0"].output_ids, eng.seqs["req-1"].output_ids]
    del eng
    torch.cuda.empty_cache()

    spec = ModelSpec("tiny-gqa", "llm", "tiny-gqa", max_model_len=256,
                     max_num_seqs=4, kv_cache_blocks=128, tp=2)
    inst = TPLLMInstance(spec, 2, device_type="cuda", backend="gloo",
                         sd_path=sd_path, start_timeout=300,
                         device_indices=[0, 0])
    try:
        for sid, prompt, exp in (("req-0", prompts[0], want[0]),
                                 ("req-1", prompts[1], want[1])):
            done = threading.Event()
            toks = []

            def cb(seq, tok, fin):
                toks.append(tok)
                if fin:
                    done.set()
            inst.submit(sid, prompt, sp, cb)
            assert done.wait(timeout=240), "stream did not finish"
            assert toks == exp, (sid, toks, exp)
    finally:
        inst.shutdown()
