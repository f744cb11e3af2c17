"""Guided decoding (OpenAI response_format JSON mode): character-level
JSON prefix automaton, token-level masking, and end-to-end engine runs.
Reference parity: the stack delegates structured outputs to vLLM's
guided decoding; engine/guided.py is the native counterpart."""

import json

import pytest
import torch

from production_stack_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    SchedulerConfig,
)
from production_stack_amd.engine.engine import LLMEngine
from production_stack_amd.engine.guided import (
    GuidedJsonState,
    JsonPrefixValidator,
    guided_state_from_response_format,
)
from production_stack_amd.engine.sampling import SamplingParams
from production_stack_amd.engine.tokenizer import BaseTokenizer


def accepts(text):
    return JsonPrefixValidator().feed_text(text)


def complete(text):
    v = JsonPrefixValidator()
    assert v.feed_text(text)
    return v.complete


def test_json_prefix_validator_accepts_valid_prefixes():
    for t in ['{', '{"a"', '{"a":', '{"a": 1', '{"a": 1,', '{"a": 1, "b',
              '[', '[1, 2', '[{"x": [true, null]}', '"hel', '"he\\"l',
              '"\\u00', '-1.5e+', '  {"a": "b"}  ', 'tru', '[]', '{}',
              '[1,2,3]', '{"a": {"b": [false]}}']:
        assert accepts(t), t


def test_json_prefix_validator_rejects_invalid():
    for t in ['{,', '{"a"1', '{"a": 1,}', '01', '1.2.3', 'tru!', '}',
              ']', '{"a" "b"}', '[1 2]', '{"a": }', '--1', '1e+}',
              '{"a": 1}}', '[1,]', '{1: 2}', '"\\x"', 'nulll']:
        v = JsonPrefixValidator()
        assert not v.feed_text(t), t


def test_json_prefix_validator_completion():
    assert complete('{"a": 1}')
    assert complete('[1, 2]')
    assert complete('"x"')
    assert complete('true')
    assert complete('42')          # bare number can end here
    assert not complete('{"a": 1')
    assert not complete('[1,')
    assert not complete('"unterminated')
    # after completion only whitespace extends the text
    v = JsonPrefixValidator()
    assert v.feed_text('{}')
    assert v.feed(' ') and not v.feed('{')


class JsonToyTokenizer(BaseTokenizer):
    """Vocab with JSON fragments on low ids, garbage words above."""

    TABLE = {3: '{', 4: '}', 5: '"a"', 6: ': ', 7: '1', 8: ', ',
             9: '"b"', 10: '[', 11: ']', 12: '2', 13: 'true',
             14: '"txt"'}

    def __init__(self, vocab_size):
        self.vocab_size = vocab_size
        self.eos_token_id = 2

    def encode(self, text):
        return [3, 5, 6, 7, 4]  # unused by the guided path

    def decode_token(self, tid):
        if tid == self.eos_token_id:
            return ""
        return self.TABLE.get(tid, f"w{tid} ")

    def decode(self, ids):
        return "".join(self.decode_token(t) for t in ids)


def test_allowed_mask_tracks_grammar():
    tok = JsonToyTokenizer(64)
    gs = GuidedJsonState()
    logits = torch.zeros(64)
    allowed, force = gs.allowed_mask(tok, logits, tok.eos_token_id)
    assert not force
    # at the start: object/array/number/string/true starters only
    assert set(allowed) <= {3, 5, 7, 9, 10, 12, 13, 14}
    assert 3 in allowed and 4 not in allowed and 8 not in allowed

    gs.v.feed_text('{')
    allowed, _ = gs.allowed_mask(tok, logits, tok.eos_token_id)
    assert set(allowed) <= {4, 5, 9, 14}  # any string key or close
    gs.v.feed_text('"a"')
    allowed, _ = gs.allowed_mask(tok, logits, tok.eos_token_id)
    assert allowed and set(allowed) <= {6}
    gs.v.feed_text(': 1')
    gs2 = gs  # after "{"a": 1" -> comma, close, or digits extending 1
    allowed, _ = gs2.allowed_mask(tok, logits, tok.eos_token_id)
    assert 4 in allowed and 8 in allowed
    gs.v.feed_text('}')
    allowed, force = gs.allowed_mask(tok, logits, tok.eos_token_id)
    assert force and allowed == [tok.eos_token_id]


def test_allowed_mask_no_json_vocab_forces_eos():
    class Wordy(BaseTokenizer):
        def __init__(self):
            self.eos_token_id = 2

        def decode_token(self, t):
            return f"w{t} "

    gs = GuidedJsonState()
    allowed, force = gs.allowed_mask(Wordy(), torch.zeros(32), 2)
    assert force and allowed == [2]


def test_guided_schema_finish_reason():
    gs = GuidedJsonState(schema={"type": "object", "required": ["name"]})
    gs.text = '{"name": "x"}'
    assert gs.finish_reason() is None
    gs.text = '{"other": 1}'
    assert gs.finish_reason() == "error_json_schema"
    gs.text = '{"broken": '
    assert gs.finish_reason() == "error_json_schema"
    assert guided_state_from_response_format(
        {"type": "json_object"}).schema is None
    assert guided_state_from_response_format({"type": "text"}) is None


def _engine():
    cfg = EngineConfig(
        model="tiny-llama",
        max_model_len=256,
        seed=3,
        cache=CacheConfig(num_gpu_blocks=64, block_size=16),
        scheduler=SchedulerConfig(max_num_seqs=4,
                                  max_num_batched_tokens=128),
    )
    eng = LLMEngine(cfg, device="cpu")
    tok = JsonToyTokenizer(eng.model_cfg.vocab_size)
    eng.tokenizer = tok
    eng.runner.tokenizer = tok
    return eng


@pytest.mark.parametrize("temperature", [0.0, 1.0])
def test_guided_json_end_to_end(temperature):
    """Every emitted token keeps the output a valid JSON prefix; a
    'stop'-finished request parses as JSON."""
    eng = _engine()
    p = SamplingParams(max_tokens=48, temperature=temperature, seed=5,
                       response_format={"type": "json_object"})
    eng.add_request("g0", [17, 18, 19], p)
    reason, toks = None, []
    for _ in range(80):
        for out in eng.step():
            if out.request_id == "g0":
                toks.extend(out.new_token_ids)
                if out.finished:
                    reason = out.finish_reason
        if reason:
            break
    assert reason in ("stop", "length")
    tok = eng.tokenizer
    text = "".join(tok.decode_token(t) for t in toks
                   if t != tok.eos_token_id)
    assert JsonPrefixValidator().feed_text(text), text
    if reason == "stop":
        json.loads(text)


def test_guided_json_async_falls_back_to_sync():
    """response_format rows must take the sync sampling path (guided
    masks need the previous token on host) and still produce valid
    JSON under async scheduling."""
    cfg = EngineConfig(
        model="tiny-llama",
        max_model_len=256,
        seed=3,
        async_scheduling=True,
        cache=CacheConfig(num_gpu_blocks=64, block_size=16),
        scheduler=SchedulerConfig(max_num_seqs=4,
                                  max_num_batched_tokens=128),
    )
    eng = LLMEngine(cfg, device="cpu")
    tok = JsonToyTokenizer(eng.model_cfg.vocab_size)
    eng.tokenizer = tok
    eng.runner.tokenizer = tok
    p = SamplingParams(max_tokens=32, temperature=0.0,
                       response_format={"type": "json_object"})
    eng.add_request("g1", [21, 22], p)
    toks, reason = [], None
    for _ in range(80):
        for out in eng.step():
            if out.request_id == "g1":
                toks.extend(out.new_token_ids)
                if out.finished:
                    reason = out.finish_reason
        if reason:
            break
    text = "".join(tok.decode_token(t) for t in toks
                   if t != tok.eos_token_id)
    assert JsonPrefixValidator().feed_text(text), text
    assert reason in ("stop", "length")


def test_schema_guide_forces_template():
    """Flat object schemas compile to a forced template: exact keys in
    order, typed values, nothing else."""
    from production_stack_amd.engine.guided import SchemaGuide

    schema = {
        "type": "object",
        "additionalProperties": False,
        "properties": {
            "name": {"type": "string"},
            "age": {"type": "number"},
            "ok": {"type": "boolean"},
        },
        "required": ["name", "age"],
    }
    g = SchemaGuide.compile(schema)
    assert g is not None
    assert g.feed_text('{"name": "bo')
    assert g.copy().feed_text('b", "age": 31, "ok": true}')
    full = '{"name": "bob", "age": 31, "ok": false}'
    h = SchemaGuide.compile(schema)
    assert h.feed_text(full) and h.complete
    # wrong key order / extra keys / wrong value type all rejected
    assert not SchemaGuide.compile(schema).feed_text('{"age": 3')
    assert not SchemaGuide.compile(schema).feed_text('{"name": 12')
    assert not SchemaGuide.compile(schema).feed_text(
        '{"name": "b", "x": 1')
    h2 = SchemaGuide.compile(schema)
    assert h2.feed_text(full)
    assert not h2.feed(',')  # nothing after the closing brace

    # nested / loose schemas fall back to generic
    assert SchemaGuide.compile({"type": "object"}) is None
    assert SchemaGuide.compile(
        {"type": "object", "properties": {"a": {"type": "string"}}}
    ) is None  # additionalProperties not false


def test_schema_guided_generation_end_to_end():
    """json_schema with a flat object: generated output IS the schema's
    shape (enforced during decoding, not just validated after)."""
    eng = _engine()
    schema = {
        "type": "object",
        "additionalProperties": False,
        "properties": {"a": {"type": "number"},
                       "b": {"type": "number"}},
        "required": ["a", "b"],
    }
    p = SamplingParams(
        max_tokens=48, temperature=1.0, seed=3,
        response_format={"type": "json_schema",
                         "json_schema": {"schema": schema}},
    )
    eng.add_request("gs", [11, 12], p)
    toks, reason = [], None
    for _ in range(80):
        for out in eng.step():
            if out.request_id == "gs":
                toks.extend(out.new_token_ids)
                if out.finished:
                    reason = out.finish_reason
        if reason:
            break
    tok = eng.tokenizer
    text = "".join(tok.decode_token(t) for t in toks
                   if t != tok.eos_token_id)
    if reason == "stop":
        obj = json.loads(text)
        assert set(obj) == {"a", "b"}
        assert all(isinstance(v, (int, float)) for v in obj.values())
    else:
        assert reason in ("length", "error_json_schema")


def test_regex_and_choice_guides():
    from production_stack_amd.engine.guided import (
        RegexGuide,
        choice_regex,
        guided_state_from_response_format,
    )

    g = RegexGuide(r"[0-9]{3}-[0-9]{4}")
    assert g.feed_text("123-")
    assert not g.complete
    assert g.feed_text("4567") and g.complete
    assert RegexGuide(r"[0-9]+").would_accept("12") is not None
    assert RegexGuide(r"[0-9]+").would_accept("a") is None
    # open-ended pattern: complete but longer matches stay legal
    h = RegexGuide(r"[0-9]+")
    assert h.feed_text("1") and h.complete
    assert h.would_accept("2") is not None

    gs = guided_state_from_response_format(
        {"type": "choice", "choices": ["yes", "no"]})
    assert gs.v.feed_text("ye") and not gs.v.complete
    assert gs.v.feed_text("s") and gs.v.complete
    assert gs.v.would_accept("x") is None
    assert choice_regex(["a.b", "c"]) == r"a\.b|c"


def test_guided_choice_end_to_end():
    """guided_choice: the model is forced onto one of the choices, then
    EOS (choices spelled in the toy vocab's token texts)."""
    eng = _engine()
    p = SamplingParams(
        max_tokens=8, temperature=1.0, seed=2,
        response_format={"type": "choice",
                         "choices": ["true", "12", "[]"]},
    )
    eng.add_request("gc", [9, 9], p)
    toks, reason = [], None
    for _ in range(40):
        for out in eng.step():
            if out.request_id == "gc":
                toks.extend(out.new_token_ids)
                if out.finished:
                    reason = out.finish_reason
        if reason:
            break
    tok = eng.tokenizer
    text = "".join(tok.decode_token(t) for t in toks
                   if t != tok.eos_token_id)
    assert reason == "stop"
    assert text in ("true", "12", "[]"), text


def test_guided_with_min_tokens_does_not_deadlock():
    """min_tokens suppresses EOS; once the grammar completes only EOS is
    legal — the grammar must win instead of producing an all--inf row."""
    eng = _engine()
    p = SamplingParams(max_tokens=24, temperature=0.0, min_tokens=20,
                       response_format={"type": "json_object"})
    eng.add_request("gm", [15, 16], p)
    reason = None
    for _ in range(60):
        for out in eng.step():
            if out.request_id == "gm" and out.finished:
                reason = out.finish_reason
        if reason:
            break
    assert reason in ("stop", "length")
