"""Property-based tests (hypothesis) for core invariants: allocator
refcounts, chunker coverage, crypto roundtrip, tokenizer identity,
GGUF metadata roundtrip."""
import string

from hypothesis import given, settings
from hypothesis import strategies as st

from helix_amd.engine.kv_cache import BlockAllocator
from helix_amd.rag.chunker import chunk_text
from helix_amd.server.crypto import decrypt_str, encrypt_str
from helix_amd.utils.tokenizer import ByteTokenizer


@settings(max_examples=60, deadline=None)
@given(st.lists(st.sampled_from(["alloc", "free", "share"]),
                min_size=1, max_size=120),
       st.integers(min_value=2, max_value=24))
def test_block_allocator_invariants(ops, nblocks):
    """Random alloc/free/share sequences: no double-allocation, counts
    conserved, freeing everything restores capacity."""
    a = BlockAllocator(nblocks)
    live = []          # blocks we hold (with multiplicity = refcount)
    for op in ops:
        if op == "alloc" and a.can_allocate(1):
            (b,) = a.allocate(1)
            live.append(b)
        elif op == "free" and live:
            b = live.pop()
            a.free([b])
        elif op == "share" and live:
            b = live[0]
            a.share(b)
            live.append(b)
        # invariant: a block's refcount equals our multiplicity
        for b in set(live):
            assert a.ref.get(b, 0) == live.count(b)
        assert a.free_count + len(set(live)) == nblocks
    a.free(list(live))
    assert a.free_count == nblocks


@settings(max_examples=60, deadline=None)
@given(st.text(alphabet=string.printable, min_size=0, max_size=4000),
       st.integers(min_value=32, max_value=512))
def test_chunker_coverage_and_bounds(text, size):
    chunks = chunk_text(text, chunk_size=size, overlap=min(16, size // 4))
    joined = "\n".join(c["text"] for c in chunks)
    # every non-whitespace word of the input appears in some chunk
    for w in text.split():
        assert w in joined or any(w in c["text"] for c in chunks) or \
            len(w) > size  # long words are window-split
    # chunks never wildly exceed the budget (word packing + overlap)
    for c in chunks:
        assert len(c["text"]) <= 2 * size + 64


@settings(max_examples=60, deadline=None)
@given(st.text(min_size=0, max_size=500),
       st.text(min_size=1, max_size=40))
def test_crypto_roundtrip_any_unicode(plaintext, key):
    assert decrypt_str(encrypt_str(plaintext, key), key) == plaintext


@settings(max_examples=60, deadline=None)
@given(st.text(min_size=0, max_size=500))
def test_byte_tokenizer_identity(text):
    t = ByteTokenizer()
    assert t.decode(t.encode(text)) == text


@settings(max_examples=30, deadline=None)
@given(st.dictionaries(
    st.text(alphabet=string.ascii_letters + ".", min_size=1, max_size=24),
    st.one_of(st.integers(min_value=0, max_value=2**31 - 1),
              st.text(max_size=40), st.booleans(),
              st.floats(min_value=-1e6, max_value=1e6,
                        allow_nan=False, width=32)),
    max_size=8))
def test_gguf_metadata_roundtrip(tmp_path_factory, meta):
    import torch

    from helix_amd.engine import gguf
    path = str(tmp_path_factory.mktemp("gguf") / "m.gguf")
    gguf.write_gguf(path, meta, {"w": torch.zeros(2, 2)})
    g = gguf.GGUFFile(path)
    for k, v in meta.items():
        got = g.metadata[k]
        if isinstance(v, float):
            assert abs(got - v) <= max(1e-3, abs(v) * 1e-5)
        else:
            assert got == v


@settings(max_examples=15, deadline=None)
@given(st.lists(
    st.one_of(
        st.tuples(st.just("add"),
                  st.integers(min_value=1, max_value=90),   # prompt len
                  st.integers(min_value=1, max_value=12)),  # max_tokens
        st.tuples(st.just("cancel"), st.integers(min_value=0, max_value=20),
                  st.just(0)),
        st.tuples(st.just("step"), st.just(0), st.just(0)),
    ), min_size=1, max_size=40))
def test_engine_fuzz_conserves_blocks(ops):
    """Random add/cancel/step interleavings: the engine never crashes,
    and when drained every block and row slot returns."""
    from helix_amd.engine.engine import EngineConfig, LLMEngine
    from helix_amd.engine.sampling_params import SamplingParams
    eng = LLMEngine(EngineConfig(model="tiny", max_model_len=128,
                                 max_num_seqs=4, kv_cache_blocks=32,
                                 max_prefill_tokens=48, eos_token_id=-1,
                                 seed=1),
                    device="cpu")
    n = 0
    for op, a, b in ops:
        if op == "add":
            n += 1
            eng.add_request(f"f{n}", [(i % 400) + 1 for i in range(a)],
                            SamplingParams(temperature=0.0, max_tokens=b,
                                           ignore_eos=True))
        elif op == "cancel":
            eng.cancel(f"f{a}")          # may or may not exist
        else:
            eng.step()
    guard = 0
    while eng.has_work:
        eng.step()
        guard += 1
        assert guard < 2000
    assert eng.num_free_blocks() == 32
    assert len(eng._free_rows) == 4


json_values = st.recursive(
    st.one_of(st.none(), st.booleans(),
              st.integers(min_value=-10**9, max_value=10**9),
              st.floats(allow_nan=False, allow_infinity=False,
                        width=32),
              st.text(max_size=20)),
    lambda children: st.one_of(
        st.lists(children, max_size=4),
        st.dictionaries(st.text(max_size=8), children, max_size=4)),
    max_leaves=12)


@settings(max_examples=120, deadline=None)
@given(st.dictionaries(st.text(max_size=8), json_values, max_size=4),
       st.sampled_from([None, (",", ":"), (", ", ": ")]),
       st.sampled_from([None, 1, 2]))
def test_json_mask_accepts_every_dumps(doc, seps, indent):
    """Property: whatever json.dumps emits, the byte automaton accepts
    byte-for-byte and reports completion (the grammar mask can never
    corner a model out of a document json.dumps could write)."""
    import json as _json

    from helix_amd.engine.json_mode import JSONByteMask
    raw = _json.dumps(doc, separators=seps, indent=indent).encode()
    m = JSONByteMask(strict_object=True)
    for b in raw:
        assert b in m.allowed_bytes(), \
            f"byte {bytes([b])!r} refused in {raw[:80]!r}"
        assert m.push_byte(b)
    assert m.complete


@settings(max_examples=60, deadline=None)
@given(st.lists(st.sampled_from(["pub", "fetch", "ack", "expire"]),
                min_size=5, max_size=40),
       st.integers(min_value=1, max_value=4))
def test_stream_bus_at_least_once(ops_seq, batch):
    """Property: any interleaving of publish/fetch/ack/lease-expiry
    delivers every published message at least once, never delivers a
    message while it is leased, and acked messages never reappear."""
    from helix_amd.server.pubsub import StreamBus
    from helix_amd.store import Store

    st_ = Store(":memory:")
    bus = StreamBus(st_, ack_wait_s=1e9)     # expiry only when forced
    published, acked, leased = set(), set(), {}
    n = 0
    for op in ops_seq:
        if op == "pub":
            n += 1
            seq = bus.publish("s", "subj", {"n": n})
            published.add(seq)
        elif op == "fetch":
            got = bus.fetch("s", "w", batch=batch)
            for m in got:
                assert m["seq"] not in leased, "delivered while leased"
                assert m["seq"] not in acked, "delivered after ack"
                leased[m["seq"]] = True
        elif op == "ack" and leased:
            seq = sorted(leased)[0]
            del leased[seq]
            bus.ack("s", "w", seq)
            acked.add(seq)
        elif op == "expire" and leased:
            # force every lease to expire
            doc = st_.get("bus_consumers", "s:w")
            if doc:
                doc["inflight"] = {k: 0.0 for k in doc["inflight"]}
                st_.put("bus_consumers", "s:w", doc)
            leased.clear()
    # drain: everything published must eventually deliver
    doc = st_.get("bus_consumers", "s:w")
    if doc:
        doc["inflight"] = {k: 0.0 for k in doc["inflight"]}
        st_.put("bus_consumers", "s:w", doc)
    seen = set(acked)
    for _ in range(len(published) + 1):
        for m in bus.fetch("s", "w", batch=50):
            seen.add(m["seq"])
            bus.ack("s", "w", m["seq"])
    assert seen >= published, f"lost {published - seen}"
    st_.close()
