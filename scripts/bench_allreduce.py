#!/usr/bin/env python3
"""Microbench: one-shot xGMI allreduce kernel latency at TP-decode
message sizes (two ranks sharing one GPU via hipIpc — RCCL cannot run
two ranks on one device, so the RCCL comparison at these sizes lives in
docs/TP_LATENCY_MODEL.md's link-bandwidth math).

Run on a GPU box:  python scripts/bench_allreduce.py
Writes JSON lines to stdout; copy results into profiles/.
"""
import json
import os
import socket
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


SIZES = [2048, 8192, 65536, 262144, 1048576, 2097152, 4194304]  # elements


def worker(rank, world, port, q):
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    import torch.distributed as dist
    torch.cuda.set_device(0)
    dist.init_process_group("gloo")
    from helix_amd import ops
    cap = 16 << 20
    handle = ops._native().ar_create(world, rank, cap)
    gathered = [None] * world
    dist.all_gather_object(gathered, handle.numpy().tobytes())
    ops._native().ar_open([torch.frombuffer(bytearray(b), dtype=torch.uint8)
                           for b in gathered])
    rows = []
    for n in SIZES:
        x = torch.randn(n, device="cuda").bfloat16()
        out = torch.empty_like(x)
        for _ in range(20):     # warmup
            ops._native().ar_allreduce(x, out)
        torch.cuda.synchronize()
        dist.barrier()
        iters = 200
        t0 = time.perf_counter()
        for _ in range(iters):
            ops._native().ar_allreduce(x, out)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        rows.append({"elems": n, "bytes": n * 2,
                     "us_per_op": round(dt * 1e6, 2),
                     "GB_s_pulled": round((world - 1) * n * 2 / dt / 1e9, 1)})
        dist.barrier()
    ops._native().ar_destroy()
    dist.destroy_process_group()
    if rank == 0:
        q.put(rows)


def worker_safe(rank, world, port, q):
    try:
        worker(rank, world, port, q)
    except Exception as e:
        q.put(f"rank {rank} error: {e}")
        raise


def main():
    ctx = torch.multiprocessing.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [ctx.Process(target=worker_safe, args=(r, 2, port, q),
                         daemon=True) for r in range(2)]
    for p in procs:
        p.start()
    rows = q.get(timeout=600)
    for p in procs:
        p.join(timeout=60)
    print(json.dumps({"bench": "one_shot_allreduce_2rank_1gpu",
                      "note": "both ranks share one MI355X (CU + HBM "
                              "contention); real TP pulls peer data over "
                              "dedicated xGMI links instead",
                      "rows": rows}, indent=2))


if __name__ == "__main__":
    sys.exit(main())
