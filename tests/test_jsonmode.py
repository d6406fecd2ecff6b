"""response_format JSON mode: validator unit tests + constrained-decode
end-to-end through the engine server (reference passes response_format
through to vLLM — api/openai/v1/chat_completions.go:350-515; here the
in-house sampler enforces it, kubeai_amd/engine/jsonmode.py)."""
import json

import pytest
from fastapi.testclient import TestClient

from kubeai_amd.engine import EngineConfig
from kubeai_amd.engine.jsonmode import JsonPrefixValidator
from kubeai_amd.engine.server import EngineServer, build_app


# ------------------------------------------------------------ validator
VALID_DOCS = [
    "{}",
    '{"a": 1}',
    '{"a": -1.5e-3, "b": [true, false, null]}',
    '{ "nested" : { "x" : [ {} , [ ] ] } }',
    '{"s": "with \\"escape\\" and \\u00e9"}',
    '{"a": 0, "b": 0.5}',
]
INVALID_PREFIXES = [
    "[1]",          # top level must be an object
    "tru",          # bare literal at top level
    '{"a" 1',       # missing colon
    '{"a": 1,,',    # double comma
    '{"a": 01',     # leading zero
    '{"a": 1}}',    # extra close
    '{"a": .5}',    # bare fraction
    '{"a": "\\x"}', # bad escape
    '{"a": 1} x',   # trailing garbage
]


@pytest.mark.parametrize("doc", VALID_DOCS)
def test_validator_accepts_valid(doc):
    v = JsonPrefixValidator()
    # every prefix of a valid doc must be accepted
    for i in range(1, len(doc) + 1):
        v2 = JsonPrefixValidator()
        assert v2.feed(doc[:i]), f"rejected prefix {doc[:i]!r}"
    assert v.feed(doc)
    assert v.complete
    json.loads(doc)  # sanity: really is valid JSON


@pytest.mark.parametrize("bad", INVALID_PREFIXES)
def test_validator_rejects_invalid(bad):
    v = JsonPrefixValidator()
    assert not v.feed(bad), f"accepted {bad!r}"


def test_validator_snapshot_restore():
    v = JsonPrefixValidator()
    assert v.feed('{"a": [1, ')
    snap = v.snapshot()
    assert not v.feed("}")  # ']' expected, not '}'
    v.restore(snap)
    assert v.feed("2]}")
    assert v.complete


# ------------------------------------------------------------ end-to-end
@pytest.fixture(scope="module")
def client():
    cfg = EngineConfig(
        model="llama-tiny", device="cpu", num_gpu_blocks=128, max_model_len=512
    )
    server = EngineServer(cfg, "json-model")
    server.start()
    server._ready.wait(timeout=60)
    app = build_app(server)
    with TestClient(app) as c:
        yield c, server
    server.stop()


def test_json_mode_output_parses(client):
    c, server = client
    tok = server.tokenizer
    # steer the random-weight model toward closing braces so the object
    # completes inside the token budget; the CONSTRAINT supplies validity
    # the constraint supplies validity; the bias supplies progress (a
    # random-weight model has no preference for JSON structure)
    bias = {
        str(tok.char_token("{")): 4.0,
        str(tok.char_token("}")): 6.0,
        str(tok.char_token('"')): 2.0,
    }
    r = c.post(
        "/v1/chat/completions",
        json={
            "messages": [{"role": "user", "content": "emit json"}],
            "max_tokens": 48,
            "temperature": 0,
            "logit_bias": bias,
            "response_format": {"type": "json_object"},
        },
    )
    assert r.status_code == 200, r.text
    text = r.json()["choices"][0]["message"]["content"]
    parsed = json.loads(text)  # must be valid JSON
    assert isinstance(parsed, dict)


def test_without_json_mode_output_is_not_json(client):
    c, _ = client
    r = c.post(
        "/v1/chat/completions",
        json={"messages": [{"role": "user", "content": "emit json"}],
              "max_tokens": 16, "temperature": 0},
    )
    text = r.json()["choices"][0]["message"]["content"]
    with pytest.raises(Exception):
        json.loads(text)


def test_bad_response_format_400(client):
    c, _ = client
    r = c.post(
        "/v1/chat/completions",
        json={"messages": [{"role": "user", "content": "x"}],
              "max_tokens": 4, "response_format": {"type": "yaml"}},
    )
    assert r.status_code == 400


# ----------------------------------------------------- schema-guided
def test_schema_validator_enforces_subset():
    from kubeai_amd.engine.jsonmode import SchemaValidator

    sch = {
        "type": "object",
        "properties": {
            "name": {"type": "string", "enum": ["alice", "bob"]},
            "age": {"type": "integer"},
            "tags": {"type": "array", "items": {"type": "string"}},
        },
        "required": ["name", "age"],
        "additionalProperties": False,
    }
    ok = SchemaValidator(sch)
    assert ok.feed('{"name": "alice", "age": 3, "tags": ["x", "y"]}')
    assert ok.complete
    assert not SchemaValidator(sch).feed('{"name": "carol')  # enum prefix
    assert not SchemaValidator(sch).feed('{"age": 3.5')      # integer
    assert not SchemaValidator(sch).feed('{"nope":')         # closed object
    held = SchemaValidator(sch)
    assert held.feed('{"name": "bob"')
    assert not held.feed('}')                                # required held
    assert not SchemaValidator(sch).feed('{"tags": {')       # array type


def test_schema_validator_refs_unions_consts():
    from kubeai_amd.engine.jsonmode import SchemaValidator

    sch = {
        "$defs": {"pt": {
            "type": "object",
            "properties": {"x": {"type": "number"}, "kind": {"const": "pt"}},
            "required": ["x", "kind"],
        }},
        "type": "object",
        "properties": {"p": {"$ref": "#/$defs/pt"},
                       "v": {"type": ["null", "string"]}},
        "required": ["p"],
    }
    v = SchemaValidator(sch)
    assert v.feed('{"p": {"kind": "pt", "x": -1.5e3}, "v": null}')
    assert v.complete
    assert not SchemaValidator(sch).feed('{"p": {"kind": "xx')
    assert not SchemaValidator(sch).feed('{"v": 4')


def test_schema_snapshot_restore_roundtrip():
    from kubeai_amd.engine.jsonmode import SchemaValidator

    sch = {"type": "object",
           "properties": {"name": {"enum": ["alice", "bob"]},
                          "age": {"type": "integer"}},
           "required": ["name", "age"]}
    v = SchemaValidator(sch)
    assert v.feed('{"name": "a')
    snap = v.snapshot()
    assert v.feed('lice", "age": 2}') and v.complete
    v.restore(snap)
    assert v.feed('lice", "age": 7}') and v.complete


def test_schema_guided_generation_http(client):
    """End to end: json_schema constrains output to parse AND conform
    (required key present, closed object, enum value)."""
    c, server = client
    tok = server.tokenizer
    bias = {
        str(tok.char_token("{")): 4.0,
        str(tok.char_token("}")): 6.0,
        str(tok.char_token('"')): 2.0,
    }
    schema = {
        "type": "object",
        "properties": {"k": {"enum": ["aa", "bb"]}},
        "required": ["k"],
        "additionalProperties": False,
    }
    r = c.post(
        "/v1/chat/completions",
        json={
            "messages": [{"role": "user", "content": "emit json"}],
            "max_tokens": 64,
            "temperature": 0,
            "logit_bias": bias,
            "response_format": {
                "type": "json_schema",
                "json_schema": {"name": "out", "schema": schema},
            },
        },
    )
    assert r.status_code == 200, r.text
    text = r.json()["choices"][0]["message"]["content"]
    obj = json.loads(text)
    assert set(obj) == {"k"} and obj["k"] in ("aa", "bb")
