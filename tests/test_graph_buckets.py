"""CUDAGraphRunner bucketing logic (engine/graph_runner.py) — the
(batch, seqlen) -> captured-graph key selection is pure host logic and
must be stable: a wrong bucket either replays a graph with stale
shapes or captures unboundedly many graphs.
"""
from helix_amd.engine.graph_runner import (BATCH_BUCKETS,
                                           MIN_LEN_BUCKET,
                                           CUDAGraphRunner)


def _runner(max_batch=512, max_model_len=8192):
    r = object.__new__(CUDAGraphRunner)   # bucket fns only touch these
    r.max_batch = max_batch
    r.max_model_len = max_model_len
    return r


def test_batch_bucket_monotone_cover():
    r = _runner()
    for n in range(1, 513):
        b = r.batch_bucket(n)
        assert b >= n
        assert b in BATCH_BUCKETS
    assert r.batch_bucket(1) == 1
    assert r.batch_bucket(65) == 96
    # clamped to max_batch even when the bucket list goes higher
    assert _runner(max_batch=48).batch_bucket(400) == 48


def test_len_bucket_pow2_from_floor():
    r = _runner()
    assert r.len_bucket(1) == MIN_LEN_BUCKET
    assert r.len_bucket(512) == 512
    assert r.len_bucket(513) == 1024
    assert r.len_bucket(4097) == 8192
    # clamped to the model's max
    assert r.len_bucket(100000) == 8192


def test_bucket_count_is_bounded():
    """Graph memory is proportional to distinct (batch, len) keys: the
    whole space for an 8k model must stay small."""
    r = _runner()
    keys = {(r.batch_bucket(n), r.len_bucket(l))
            for n in range(1, 513, 7) for l in range(1, 8193, 131)}
    assert len(keys) <= len(BATCH_BUCKETS) * 5
