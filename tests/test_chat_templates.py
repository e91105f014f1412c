"""Chat templates + hermes tool-call parsing (reference profile serves
with --tool-call-parser hermes, design/2026-04-28:58-60)."""
import json

from helix_amd.utils.chat_templates import (parse_tool_calls,
                                            render_chatml, render_llama3,
                                            render_mistral,
                                            template_for_model)

MSGS = [
    {"role": "system", "content": "You are helpful."},
    {"role": "user", "content": "Hi"},
    {"role": "assistant", "content": "Hello!"},
    {"role": "user", "content": "What is 2+2?"},
]


def test_llama3_template():
    t = render_llama3(MSGS)
    assert t.startswith("<|begin_of_text|><|start_header_id|>system")
    assert "You are helpful.<|eot_id|>" in t
    assert t.count("<|start_header_id|>user<|end_header_id|>") == 2
    assert t.endswith("<|start_header_id|>assistant<|end_header_id|>\n\n")


def test_mistral_template_folds_system():
    t = render_mistral(MSGS)
    assert t.startswith("<s>[INST] You are helpful.")
    assert "[INST] What is 2+2? [/INST]" in t
    assert " Hello!</s>" in t


def test_chatml_template():
    t = render_chatml(MSGS)
    assert t.startswith("<|im_start|>system\nYou are helpful.<|im_end|>")
    assert t.endswith("<|im_start|>assistant\n")


def test_template_selection():
    assert template_for_model("llama3-8b") == "llama3"
    assert template_for_model("mistral-7b") == "mistral"
    assert template_for_model("qwen2-7b") == "chatml"


def test_tools_preamble_injected():
    tools = [{"type": "function", "function": {
        "name": "get_weather", "description": "Weather lookup",
        "parameters": {"type": "object",
                       "properties": {"city": {"type": "string"}}}}}]
    t = render_llama3(MSGS, tools=tools)
    assert "<tools>" in t and "get_weather" in t
    assert "<tool_call>" in t  # instructions mention the format


def test_parse_tool_calls_single():
    text = ('Sure, checking.\n<tool_call>{"name": "get_weather", '
            '"arguments": {"city": "Oslo"}}</tool_call>')
    content, calls = parse_tool_calls(text)
    assert content == "Sure, checking."
    assert len(calls) == 1
    assert calls[0]["function"]["name"] == "get_weather"
    assert json.loads(calls[0]["function"]["arguments"]) == {"city": "Oslo"}
    assert calls[0]["id"].startswith("call_")


def test_parse_tool_calls_multiple_and_malformed():
    text = ('<tool_call>{"name": "a", "arguments": {}}</tool_call>'
            '<tool_call>{not json}</tool_call>'
            '<tool_call>{"name": "b", "arguments": {"x": 1}}</tool_call>')
    content, calls = parse_tool_calls(text)
    assert [c["function"]["name"] for c in calls] == ["a", "b"]
    assert "{not json}" in content   # malformed block left in content


def test_byte_tokenizer_template_with_tools_roundtrip():
    from helix_amd.utils.tokenizer import ByteTokenizer
    tok = ByteTokenizer()
    tools = [{"function": {"name": "calc", "parameters": {}}}]
    ids = tok.apply_chat_template(MSGS, tools=tools)
    text = tok.decode(ids)
    assert "calc" in text and "You are helpful." in text


def test_hf_tokenizer_uses_family_template(tmp_path):
    """A real tokenizer.json (trained offline on the fly) encodes the
    llama3-format rendering losslessly."""
    from tokenizers import Tokenizer, models, pre_tokenizers, trainers
    tok = Tokenizer(models.BPE(unk_token="<unk>"))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    trainer = trainers.BpeTrainer(
        vocab_size=600,
        special_tokens=["<unk>", "<|begin_of_text|>", "<|end_of_text|>",
                        "<|start_header_id|>", "<|end_header_id|>",
                        "<|eot_id|>"])
    tok.train_from_iterator(
        ["You are helpful. Hi Hello! What is 2+2? system user assistant"
         " weather city json name arguments"] * 50, trainer)
    path = str(tmp_path / "tokenizer.json")
    tok.save(path)

    from helix_amd.utils.tokenizer import HFTokenizer
    h = HFTokenizer(path)
    ids = h.apply_chat_template(MSGS, template="llama3")
    assert ids[0] == h._tok.token_to_id("<|begin_of_text|>")
    assert h._tok.token_to_id("<|eot_id|>") in ids


def test_adapter_nonstream_tool_calls(monkeypatch):
    """chat_completion surfaces hermes tool calls as OpenAI tool_calls
    with finish_reason=tool_calls (generation stubbed)."""
    import asyncio
    from helix_amd.runner import openai_adapter as oa

    class FakeSpec:
        max_model_len = 512

    class FakeInst(oa.LLMInstance):
        def __init__(self):
            self.spec = FakeSpec()

    class FakeSvc:
        def ensure_loaded(self, model):
            return FakeInst()

    async def fake_gen(inst, seq_id, prompt_ids, params, stop, tok, loop):
        return ('<tool_call>{"name": "calc", "arguments": {"a": 2}}'
                '</tool_call>', "stop", 12, [])

    monkeypatch.setattr(oa, "_generate_one", fake_gen)
    req = {"model": "llama3-8b",
           "messages": [{"role": "user", "content": "2+2?"}],
           "tools": [{"type": "function",
                      "function": {"name": "calc", "parameters": {}}}]}
    resp = asyncio.run(oa.chat_completion(FakeSvc(), req))
    ch = resp["choices"][0]
    assert ch["finish_reason"] == "tool_calls"
    assert ch["message"]["tool_calls"][0]["function"]["name"] == "calc"


def test_hf_jinja_chat_template(tmp_path, monkeypatch):
    """A model shipping its own chat_template (HF tokenizer_config
    jinja) takes precedence over the built-in family templates."""
    import json

    from helix_amd.utils.tokenizer import get_tokenizer

    mdir = tmp_path / "jinja-model"
    mdir.mkdir()
    # minimal real tokenizer.json (WordLevel with a tiny vocab)
    vocab = {c: i for i, c in enumerate(
        ["<s>", "</s>", "[UNK]", "user", "assistant", ":", "hi",
         "there", "SYS", "\n", " "])}
    (mdir / "tokenizer.json").write_text(json.dumps({
        "version": "1.0",
        "truncation": None, "padding": None,
        "added_tokens": [], "normalizer": None,
        "pre_tokenizer": {"type": "Whitespace"},
        "post_processor": None, "decoder": None,
        "model": {"type": "WordLevel", "vocab": vocab,
                  "unk_token": "[UNK]"}}))
    (mdir / "tokenizer_config.json").write_text(json.dumps({
        "bos_token": "<s>", "eos_token": "</s>",
        "chat_template":
            "{{ bos_token }}{% for m in messages %}"
            "{{ m.role }} : {{ m.content }}\n{% endfor %}"
            "{% if add_generation_prompt %}assistant :{% endif %}"}))
    monkeypatch.setenv("HELIX_TOKENIZER_DIR", str(tmp_path))
    tok = get_tokenizer("jinja-model")
    assert tok.chat_template is not None
    ids = tok.apply_chat_template(
        [{"role": "user", "content": "hi there"}],
        add_generation_prompt=True)
    text_tokens = [k for k, v in sorted(vocab.items(),
                                        key=lambda kv: kv[1])]
    rendered = " ".join(text_tokens[i] for i in ids)
    assert "user" in rendered and "hi" in rendered
    assert rendered.rstrip().endswith("assistant :")
