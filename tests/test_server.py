"""Engine HTTP server: the full engine contract from SURVEY.md §2.16-bis.

Uses the tiny model on CPU with FastAPI's TestClient (no sockets).
"""
import json

import pytest
from fastapi.testclient import TestClient

from kubeai_amd.engine.engine import EngineConfig
from kubeai_amd.engine.server import EngineServer, build_app


@pytest.fixture(scope="module")
def client():
    cfg = EngineConfig(
        model="llama-tiny", device="cpu", num_gpu_blocks=256, max_model_len=512
    )
    server = EngineServer(cfg, "test-model")
    server.start()
    server._ready.wait(timeout=60)
    app = build_app(server)
    with TestClient(app) as c:
        yield c
    server.stop()


def test_health(client):
    r = client.get("/health")
    assert r.status_code == 200
    assert r.json()["status"] == "ok"


def test_completions_usage(client):
    r = client.post(
        "/v1/completions",
        json={"model": "test-model", "prompt": "hello world foo bar", "max_tokens": 5,
              "temperature": 0},
    )
    assert r.status_code == 200
    body = r.json()
    assert body["choices"][0]["finish_reason"] in ("length", "stop")
    u = body["usage"]
    # k6 reads these fields (benchmarks/multi-turn-chat-k6/k6.js:52-53)
    assert u["prompt_tokens"] == 5  # bos + 4 words
    assert 1 <= u["completion_tokens"] <= 5
    assert u["total_tokens"] == u["prompt_tokens"] + u["completion_tokens"]


def test_chat_completions(client):
    r = client.post(
        "/v1/chat/completions",
        json={
            "model": "test-model",
            "messages": [
                {"role": "system", "content": "be brief"},
                {"role": "user", "content": "hi there"},
            ],
            "max_tokens": 4,
            "temperature": 0,
        },
    )
    assert r.status_code == 200
    body = r.json()
    assert body["object"] == "chat.completion"
    assert body["choices"][0]["message"]["role"] == "assistant"
    assert body["usage"]["completion_tokens"] >= 1


def test_chat_streaming(client):
    with client.stream(
        "POST",
        "/v1/chat/completions",
        json={
            "model": "test-model",
            "messages": [{"role": "user", "content": "stream me"}],
            "max_tokens": 4,
            "temperature": 0,
            "stream": True,
        },
    ) as r:
        assert r.status_code == 200
        events = []
        for line in r.iter_lines():
            if line.startswith("data: "):
                events.append(line[len("data: ") :])
    assert events[-1] == "[DONE]"
    payloads = [json.loads(e) for e in events[:-1]]
    assert payloads[0]["object"] == "chat.completion.chunk"
    assert any(p["choices"][0]["finish_reason"] for p in payloads)
    assert "usage" in payloads[-1]


def test_models_listing(client):
    r = client.get("/v1/models")
    ids = [m["id"] for m in r.json()["data"]]
    assert "test-model" in ids


def test_metrics_exposition(client):
    # generate some load first
    client.post(
        "/v1/completions",
        json={"prompt": "metrics probe", "max_tokens": 2, "temperature": 0},
    )
    r = client.get("/metrics")
    text = r.text
    assert "kubeai_engine_num_requests_waiting" in text
    assert "kubeai_engine_kv_cache_usage_perc" in text
    assert "kubeai_engine_generation_tokens_total" in text
    assert "kubeai_engine_time_to_first_token_seconds" in text


def test_lora_admin_semantics(client):
    # load
    r = client.post(
        "/v1/load_lora_adapter", json={"lora_name": "ad1", "lora_path": None}
    )
    assert r.status_code == 200
    # double load -> "already loaded" (vllmclient matches this string,
    # internal/vllmclient/client.go:30-45)
    r = client.post(
        "/v1/load_lora_adapter", json={"lora_name": "ad1", "lora_path": None}
    )
    assert r.status_code == 400 and "already loaded" in r.text
    # adapter appears in /v1/models and is usable as model name
    ids = [m["id"] for m in client.get("/v1/models").json()["data"]]
    assert "ad1" in ids
    r = client.post(
        "/v1/completions",
        json={"model": "ad1", "prompt": "adapter run", "max_tokens": 3,
              "temperature": 0},
    )
    assert r.status_code == 200
    # unload
    r = client.post("/v1/unload_lora_adapter", json={"lora_name": "ad1"})
    assert r.status_code == 200
    # unload again -> "cannot be found" (client.go:59-73)
    r = client.post("/v1/unload_lora_adapter", json={"lora_name": "ad1"})
    assert r.status_code == 404 and "cannot be found" in r.text


def test_lora_changes_output(client):
    base = client.post(
        "/v1/completions",
        json={"prompt": "same prompt here", "max_tokens": 4, "temperature": 0},
    ).json()["choices"][0]["text"]
    client.post("/v1/load_lora_adapter", json={"lora_name": "delta", "lora_path": None})
    adapted = client.post(
        "/v1/completions",
        json={"model": "delta", "prompt": "same prompt here", "max_tokens": 4,
              "temperature": 0},
    ).json()["choices"][0]["text"]
    client.post("/v1/unload_lora_adapter", json={"lora_name": "delta"})
    # random-init adapter must perturb the greedy path
    assert adapted != base


def test_n_choices(client):
    r = client.post(
        "/v1/completions",
        json={"prompt": "pick some words", "max_tokens": 4, "n": 3,
              "temperature": 0.9, "ignore_eos": True},
    )
    assert r.status_code == 200
    body = r.json()
    assert len(body["choices"]) == 3
    assert [c["index"] for c in body["choices"]] == [0, 1, 2]
    # temperature sampling with distinct auto-seeds -> diverse outputs
    texts = {c["text"] for c in body["choices"]}
    assert len(texts) >= 2
    assert body["usage"]["completion_tokens"] == 12


def test_stop_strings(client):
    base = client.post(
        "/v1/completions",
        json={"prompt": "stop string probe", "max_tokens": 8, "temperature": 0},
    ).json()["choices"][0]["text"]
    words = base.split()
    assert len(words) >= 3
    stop_word = words[2]
    r = client.post(
        "/v1/completions",
        json={"prompt": "stop string probe", "max_tokens": 8, "temperature": 0,
              "stop": [stop_word]},
    ).json()
    choice = r["choices"][0]
    assert choice["finish_reason"] == "stop"
    assert stop_word not in choice["text"]
    assert choice["text"].split() == words[:2]


def test_penalties_change_output(client):
    base = client.post(
        "/v1/completions",
        json={"prompt": "penalty probe text", "max_tokens": 10,
              "temperature": 0},
    ).json()["choices"][0]["text"]
    pen = client.post(
        "/v1/completions",
        json={"prompt": "penalty probe text", "max_tokens": 10,
              "temperature": 0, "frequency_penalty": 2.0,
              "presence_penalty": 2.0},
    ).json()["choices"][0]["text"]
    # strong penalties must forbid immediate repeats; outputs diverge once
    # the greedy path would have repeated a token
    base_toks = base.split()
    pen_toks = pen.split()
    if len(set(base_toks)) < len(base_toks):  # base repeats something
        assert pen != base
    assert len(set(pen_toks)) == len(pen_toks)  # no repeats under penalty


def test_unknown_model_404(client):
    r = client.post(
        "/v1/completions", json={"model": "nope", "prompt": "x", "max_tokens": 1}
    )
    assert r.status_code == 404


def test_rerank(client):
    r = client.post(
        "/v1/rerank",
        json={
            "model": "test-model",
            "query": "alpha beta",
            "documents": ["alpha beta gamma", "unrelated words here", "alpha beta"],
            "top_n": 2,
        },
    )
    assert r.status_code == 200
    results = r.json()["results"]
    assert len(results) == 2
    # the identical-prefix documents must outrank the unrelated one
    assert {res["index"] for res in results} <= {0, 1, 2}
    assert results[0]["relevance_score"] >= results[1]["relevance_score"]
    assert 1 not in {res["index"] for res in results}


def test_audio_transcription_not_supported(client):
    r = client.post("/v1/audio/transcriptions")
    assert r.status_code == 501


def test_embeddings(client):
    r = client.post(
        "/v1/embeddings",
        json={"model": "test-model", "input": ["hello world", "other text"]},
    )
    assert r.status_code == 200
    data = r.json()["data"]
    assert len(data) == 2
    v = data[0]["embedding"]
    assert len(v) == 256  # llama-tiny hidden size
    norm = sum(x * x for x in v) ** 0.5
    assert abs(norm - 1.0) < 1e-3


def test_logprobs(client):
    r = client.post(
        "/v1/completions",
        json={"prompt": "log probs please", "max_tokens": 4, "temperature": 0,
              "logprobs": 2},
    )
    assert r.status_code == 200
    obj = r.json()["choices"][0]["logprobs"]
    lp = obj["token_logprobs"]
    assert len(lp) == 4
    assert all(v <= 0 for v in lp)
    assert len(obj["tokens"]) == 4
    # 2 alternatives per position; greedy choice == best alternative
    assert all(len(top) == 2 for top in obj["top_logprobs"])
    for chosen_lp, top in zip(lp, obj["top_logprobs"]):
        assert abs(max(top.values()) - chosen_lp) < 1e-4


def test_chat_logprobs(client):
    r = client.post(
        "/v1/chat/completions",
        json={"model": "test-model",
              "messages": [{"role": "user", "content": "chat logprobs"}],
              "max_tokens": 3, "temperature": 0,
              "logprobs": True, "top_logprobs": 2},
    )
    assert r.status_code == 200
    content = r.json()["choices"][0]["logprobs"]["content"]
    assert len(content) == 3
    for item in content:
        assert item["logprob"] <= 0
        assert len(item["top_logprobs"]) == 2
        assert item["top_logprobs"][0]["logprob"] >= item["top_logprobs"][1]["logprob"]


def test_logit_bias_forces_token(client):
    # +100 bias on one token id dominates every step (greedy)
    forced = 777
    r = client.post(
        "/v1/completions",
        json={"prompt": "bias probe", "max_tokens": 3, "temperature": 0,
              "logit_bias": {str(forced): 100.0}},
    )
    assert r.status_code == 200
    # re-encode the text through the synthetic tokenizer: every generated
    # token must be the forced id
    text = r.json()["choices"][0]["text"]
    assert r.json()["usage"]["completion_tokens"] == 3
    # direct engine-level check of the sampled ids
    r2 = client.post(
        "/v1/completions",
        json={"prompt": "bias probe", "max_tokens": 3, "temperature": 0,
              "logit_bias": {str(forced): 100.0}, "logprobs": 1},
    )
    lp = r2.json()["choices"][0]["logprobs"]
    assert len(lp["tokens"]) == 3
    assert len(set(lp["tokens"])) == 1  # same forced token every step


def test_logit_bias_out_of_range_400(client):
    # an out-of-range logit_bias id must be a clean 400
    # at parse time, never a step-loop crash
    r = client.post(
        "/v1/completions",
        json={"prompt": "oob bias", "max_tokens": 2, "temperature": 0,
              "logit_bias": {"999999999": 5.0}},
    )
    assert r.status_code == 400
    assert "logit_bias" in r.json()["error"]["message"]
    r = client.post(
        "/v1/chat/completions",
        json={"messages": [{"role": "user", "content": "x"}],
              "max_tokens": 2, "logit_bias": {"-5": 1.0}},
    )
    assert r.status_code == 400
    # engine still alive afterwards
    ok = client.post(
        "/v1/completions",
        json={"prompt": "still alive", "max_tokens": 2, "temperature": 0},
    )
    assert ok.status_code == 200
    assert client.get("/health").status_code == 200


def test_step_crash_aborts_requests_and_recovers(client):
    # force one step() exception and verify: the in-flight request gets a
    # finished(abort) response instead of hanging, /health recovers, and
    # the next request works
    eng_server = client.app.state.eng_server
    real_step = eng_server.engine.step
    calls = {"n": 0}

    def boom():
        if calls["n"] == 0:
            calls["n"] += 1
            raise RuntimeError("injected step failure")
        return real_step()

    eng_server.engine.step = boom
    try:
        r = client.post(
            "/v1/completions",
            json={"prompt": "crash probe", "max_tokens": 4, "temperature": 0},
        )
        # request completed (aborted with empty text) rather than hanging
        assert r.status_code == 200
        assert r.json()["choices"][0]["finish_reason"] in ("abort", "stop")
    finally:
        eng_server.engine.step = real_step
    ok = client.post(
        "/v1/completions",
        json={"prompt": "recovery probe", "max_tokens": 2, "temperature": 0},
    )
    assert ok.status_code == 200
    assert ok.json()["usage"]["completion_tokens"] == 2
    assert client.get("/health").status_code == 200


def test_streaming_stop_holdback(client):
    """A stop string spanning chunk boundaries must never leak its leading
    characters into the streamed deltas."""
    base = client.post(
        "/v1/completions",
        json={"prompt": "holdback probe", "max_tokens": 8, "temperature": 0},
    ).json()["choices"][0]["text"]
    words = base.split()
    assert len(words) >= 4
    # stop spans a token boundary: last chars of word2 + separator + word3
    stop = words[2][-1] + " " + words[3]
    with client.stream(
        "POST",
        "/v1/completions",
        json={"prompt": "holdback probe", "max_tokens": 8, "temperature": 0,
              "stop": [stop], "stream": True},
    ) as r:
        body = "".join(chunk for chunk in r.iter_text())
    import json as _json

    pieces = []
    finish = None
    for line in body.splitlines():
        if line.startswith("data: ") and line != "data: [DONE]":
            c = _json.loads(line[6:])["choices"][0]
            pieces.append(c.get("text") or "")
            finish = c["finish_reason"] or finish
    text = "".join(pieces)
    assert finish == "stop"
    assert stop not in text
    expected = base[: base.find(stop)]
    assert text == expected


def test_streamed_text_matches_nonstream(client):
    js = {"prompt": "equivalence probe", "max_tokens": 8, "temperature": 0}
    want = client.post("/v1/completions", json=js).json()["choices"][0]["text"]
    with client.stream(
        "POST", "/v1/completions", json={**js, "stream": True}
    ) as r:
        body = "".join(chunk for chunk in r.iter_text())
    import json as _json

    got = "".join(
        (_json.loads(l[6:])["choices"][0].get("text") or "")
        for l in body.splitlines()
        if l.startswith("data: ") and l != "data: [DONE]"
    )
    assert got == want


def test_stream_n_gt_1(client):
    import json as _json

    with client.stream(
        "POST",
        "/v1/completions",
        json={"prompt": "stream n probe", "max_tokens": 6, "temperature": 0,
              "n": 3, "stream": True},
    ) as r:
        body = "".join(r.iter_text())
    texts = {0: "", 1: "", 2: ""}
    finishes = {}
    usage = None
    for line in body.splitlines():
        if line.startswith("data: ") and line != "data: [DONE]":
            c = _json.loads(line[6:])
            ch = c["choices"][0]
            texts[ch["index"]] += ch.get("text") or ""
            if ch["finish_reason"]:
                finishes[ch["index"]] = ch["finish_reason"]
            usage = c.get("usage") or usage
    want = client.post(
        "/v1/completions",
        json={"prompt": "stream n probe", "max_tokens": 6, "temperature": 0},
    ).json()["choices"][0]["text"]
    # greedy: all three choices identical and equal to non-stream
    assert texts[0] == texts[1] == texts[2] == want
    assert set(finishes) == {0, 1, 2}
    assert usage["completion_tokens"] == 18  # 3 choices x 6 tokens


def test_completions_echo(client):
    prompt = "echo probe words"
    base = client.post(
        "/v1/completions",
        json={"prompt": prompt, "max_tokens": 4, "temperature": 0},
    ).json()["choices"][0]["text"]
    r = client.post(
        "/v1/completions",
        json={"prompt": prompt, "max_tokens": 4, "temperature": 0, "echo": True},
    ).json()["choices"][0]["text"]
    assert r == prompt + base


def test_completions_suffix_rejected(client):
    r = client.post(
        "/v1/completions",
        json={"prompt": "x", "suffix": "y", "max_tokens": 2},
    )
    assert r.status_code == 400
    assert "suffix" in r.json()["error"]["message"]


def test_tools_roundtrip(client):
    """tools + forced tool_choice: output is either a parsed tool_call or
    (at minimum) valid JSON content; request surface accepted."""
    import json as _json

    eng_server = client.app.state.eng_server
    tok = eng_server.tokenizer
    bias = {
        str(tok.char_token("{")): 4.0,
        str(tok.char_token("}")): 6.0,
        str(tok.char_token('"')): 2.0,
    }
    tools = [{
        "type": "function",
        "function": {
            "name": "get_weather",
            "description": "get weather",
            "parameters": {"type": "object",
                           "properties": {"city": {"type": "string"}}},
        },
    }]
    r = client.post(
        "/v1/chat/completions",
        json={"messages": [{"role": "user", "content": "weather in Oslo"}],
              "max_tokens": 48, "temperature": 0, "logit_bias": bias,
              "tools": tools, "tool_choice": "required"},
    )
    assert r.status_code == 200, r.text
    choice = r.json()["choices"][0]
    if choice["finish_reason"] == "tool_calls":
        call = choice["message"]["tool_calls"][0]["function"]
        assert call["name"] == "get_weather"
        _json.loads(call["arguments"])
    else:
        # random-weight model rarely names the function; the constraint
        # still guarantees valid JSON content
        _json.loads(choice["message"]["content"])


def test_parse_tool_call_shapes():
    from kubeai_amd.engine.server import _parse_tool_call

    tools = [{"type": "function", "function": {"name": "f"}}]
    assert _parse_tool_call('{"name": "f", "arguments": {"a": 1}}', tools) == (
        "f", {"a": 1})
    assert _parse_tool_call('{"name": "g", "arguments": {}}', tools) is None
    assert _parse_tool_call("not json", tools) is None
    assert _parse_tool_call('{"name": "f"}', tools) == ("f", {})


def test_tokenize_detokenize_endpoints(client):
    r = client.post("/tokenize", json={"prompt": "hello world"})
    assert r.status_code == 200
    body = r.json()
    assert body["count"] == len(body["tokens"]) > 0
    assert body["max_model_len"] == 512
    r2 = client.post("/detokenize", json={"tokens": body["tokens"]})
    assert r2.status_code == 200
    assert isinstance(r2.json()["prompt"], str)
    r3 = client.post("/tokenize", json={
        "messages": [{"role": "user", "content": "hi"}]})
    assert r3.status_code == 200 and r3.json()["count"] > 0
    assert client.post("/tokenize", json={}).status_code == 400


def test_server_cli_quantization_flag():
    import subprocess
    import sys

    out = subprocess.run(
        [sys.executable, "-m", "kubeai_amd.engine.server", "--help"],
        capture_output=True, text=True,
    )
    assert "--quantization" in out.stdout
