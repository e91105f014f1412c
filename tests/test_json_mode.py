"""Grammar-constrained JSON decoding (OpenAI response_format
json_object; the reference serves this via its backend's guided
decoding): the byte-level automaton makes invalid JSON unrepresentable
and forces EOS once the document closes.
"""
import json

import pytest

from helix_amd.engine.json_mode import JSONByteMask


def test_automaton_accepts_valid_rejects_invalid():
    def run(doc, strict=True):
        m = JSONByteMask(strict_object=strict)
        for b in doc:
            if not m.push_byte(b):
                return None
        return m

    m = run(b'{"a": [1, 2.5, -3e2], "b": {"c": null}, "d": "x\\n"}')
    assert m is not None and m.complete
    assert run(b'{"a":}') is None
    assert run(b'{"a" 1}') is None
    assert run(b'{"a": tru}') is None
    assert run(b'[1]', strict=True) is None       # json_object => object
    m = run(b'[1, "two"]', strict=False)
    assert m is not None and m.complete
    # nothing is allowed after completion
    m = run(b'{}')
    assert m.complete and m.allowed_bytes() == set()


def test_allowed_bytes_are_exact():
    m = JSONByteMask()
    assert m.allowed_bytes() <= set(b" \t\n\r{")
    for b in b'{"k"':
        assert m.push_byte(b)
    assert m.allowed_bytes() <= set(b" \t\n\r:")
    m.push_byte(ord(":"))
    allowed = m.allowed_bytes()
    assert ord("{") in allowed and ord('"') in allowed \
        and ord("5") in allowed
    assert ord("}") not in allowed                 # value required


def test_engine_output_is_always_a_valid_json_prefix():
    """A random-init model forced through the mask can only emit bytes
    that keep the output a valid JSON prefix (full closure is a
    likelihood question, not a grammar one, with random weights)."""
    from helix_amd.engine.engine import EngineConfig, LLMEngine
    from helix_amd.engine.sampling_params import SamplingParams
    from helix_amd.utils.tokenizer import N_SPECIAL

    eng = LLMEngine(EngineConfig(model="tiny", max_num_seqs=2,
                                 max_model_len=256, kv_cache_blocks=128,
                                 eos_token_id=2), device="cpu")
    for seed in (5, 11):
        sp = SamplingParams(temperature=0.9, seed=seed, max_tokens=60,
                            json_mode=True)
        out = eng.generate([[1, 3, seed]], sp)[0]
        assert out, "no tokens generated"
        checker = JSONByteMask()
        for pos, t in enumerate(out):
            if t == 2:                    # EOS: only after completion
                assert checker.complete, "EOS before the doc closed"
                assert pos == len(out) - 1
                body = bytes(x - N_SPECIAL for x in out[:pos])
                assert isinstance(json.loads(body), dict)
                break
            b = t - N_SPECIAL
            assert 0 <= b < 256, f"non-byte token {t} escaped the mask"
            assert checker.push_byte(b), \
                f"grammar violation at byte {b!r}"


def test_completed_document_forces_eos():
    """Once the automaton reports complete, the mask admits only EOS
    and the engine finishes the sequence."""
    import torch

    from helix_amd.engine.engine import (EngineConfig, LLMEngine,
                                         Sequence)
    from helix_amd.engine.sampling_params import SamplingParams

    eng = LLMEngine(EngineConfig(model="tiny", max_num_seqs=2,
                                 max_model_len=64, kv_cache_blocks=32,
                                 eos_token_id=2), device="cpu")
    s = Sequence(seq_id="j1", prompt_ids=[1, 3],
                 params=SamplingParams(temperature=1.0, seed=3,
                                       json_mode=True))
    m = JSONByteMask()
    for b in b'{"a": 1}':
        assert m.push_byte(b)
    assert m.complete
    s.json_mask = m
    logits = torch.randn(1, 512)
    toks = eng._sample([s], logits)
    assert toks == [2]                     # forced EOS


def test_adapter_maps_response_format():
    from helix_amd.runner.openai_adapter import _params_from_request
    p, _ = _params_from_request(
        {"response_format": {"type": "json_object"},
         "max_tokens": 9}, 2048)
    assert p.json_mode is True
    p, _ = _params_from_request({"max_tokens": 9}, 2048)
    assert p.json_mode is False


def test_logit_bias():
    """OpenAI logit_bias: additive per-token bias before sampling
    (a large positive bias makes the token deterministic)."""
    import torch

    from helix_amd.engine.engine import EngineConfig, LLMEngine
    from helix_amd.engine.sampling_params import SamplingParams

    eng = LLMEngine(EngineConfig(model="tiny", max_num_seqs=2,
                                 max_model_len=64, kv_cache_blocks=32,
                                 eos_token_id=-1), device="cpu")
    sp = SamplingParams(temperature=0.0, max_tokens=4, ignore_eos=True,
                        logit_bias={77: 1000.0})
    out = eng.generate([[1, 2, 3]], sp)[0]
    assert out == [77, 77, 77, 77]
    # adapter mapping (string keys, OpenAI wire form)
    from helix_amd.runner.openai_adapter import _params_from_request
    p, _ = _params_from_request({"logit_bias": {"42": -5}}, 128)
    assert p.logit_bias == {42: -5.0}
