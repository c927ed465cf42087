"""HTTP serving front: FastAPI over the continuous batcher. The
reference's HTTP surface is dead (tests/test_server.py targets a
server.py absent from its tree); this one works and is tested."""
import pytest
import torch

from distributedllm_amd.engine import TorchSliceEngine
from distributedllm_amd.engine.tokenizer import Tokenizer
from distributedllm_amd.formats import slicer, synthetic
from distributedllm_amd.serving import ContinuousBatcher, build_http_app


@pytest.fixture()
def client_and_worker():
    from fastapi.testclient import TestClient
    f = synthetic.build_model("tiny", seed=0)
    eng = TorchSliceEngine.from_ggml(f, n_ctx=64, max_batch=2)
    eng.attach_extra(slicer.make_extra_layers(f))
    bat = ContinuousBatcher(eng)
    app, worker = build_http_app(bat, Tokenizer(f.vocab))
    with TestClient(app) as c:
        yield c, worker
    worker.stop()


def test_health(client_and_worker):
    c, _ = client_and_worker
    r = c.get("/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"


def test_generate_greedy_deterministic(client_and_worker):
    c, _ = client_and_worker
    body = {"prompt": "hello", "num_tokens": 5}
    a = c.post("/generate", json=body).json()
    b = c.post("/generate", json=body).json()
    assert len(a["tokens"]) == 5 and a["tokens"] == b["tokens"]
    assert isinstance(a["text"], str)


def test_concurrent_requests_batch(client_and_worker):
    """Two parallel requests share decode steps and both finish."""
    import threading
    c, _ = client_and_worker
    out = {}

    def go(name, prompt):
        out[name] = c.post("/generate", json={
            "prompt": prompt, "num_tokens": 6}).json()

    ts = [threading.Thread(target=go, args=(n, p))
          for n, p in (("a", "one"), ("b", "two three"))]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=60)
    assert len(out) == 2
    assert all(len(v["tokens"]) == 6 for v in out.values())


def test_sampled_generation_and_validation(client_and_worker):
    c, _ = client_and_worker
    r = c.post("/generate", json={"prompt": "x", "num_tokens": 4,
                                  "temperature": 0.8, "seed": 3,
                                  "top_k": 40})
    assert r.status_code == 200 and len(r.json()["tokens"]) == 4
    assert c.post("/generate", json={"prompt": "x",
                                     "num_tokens": 0}).status_code == 422


def test_streaming_generation(client_and_worker):
    """SSE stream: one data event per token, final done event carries
    the full text, and the stream equals the non-streaming result."""
    import json
    c, _ = client_and_worker
    want = c.post("/generate", json={"prompt": "hello",
                                     "num_tokens": 5}).json()
    pieces, final = [], None
    with c.stream("POST", "/generate_stream",
                  json={"prompt": "hello", "num_tokens": 5}) as r:
        assert r.status_code == 200
        event = None
        for line in r.iter_lines():
            if line.startswith("event:"):
                event = line.split(":", 1)[1].strip()
            elif line.startswith("data:"):
                d = json.loads(line.split(":", 1)[1])
                if event == "done":
                    final = d
                else:
                    pieces.append(d["token"])
                    event = None
    assert pieces == want["tokens"]
    assert final is not None and final["tokens"] == want["tokens"]
    assert final["text"] == want["text"]


def test_metrics_endpoint(client_and_worker):
    c, _ = client_and_worker
    c.post("/generate", json={"prompt": "m", "num_tokens": 3})
    m = c.get("/metrics").json()
    assert m["requests_served"] >= 1 and m["tokens_generated"] >= 3
    assert m["slots"] >= 1 and m["lanes"] == 1


def test_stop_strings(client_and_worker):
    """`stop` ends generation at the first occurrence of a stop string,
    excluded from the returned text (OpenAI semantics)."""
    c, _ = client_and_worker
    full = c.post("/generate",
                  json={"prompt": "hello", "num_tokens": 12}).json()
    assert len(full["tokens"]) == 12
    # use a piece from the middle of the greedy continuation as stop
    mid = len(full["text"]) // 2
    stop = full["text"][mid:mid + 2]
    assert stop and stop in full["text"]
    r = c.post("/generate", json={"prompt": "hello", "num_tokens": 12,
                                  "stop": [stop]}).json()
    assert stop not in r["text"]
    assert full["text"].startswith(r["text"])
    assert len(r["tokens"]) < 12
    # a stop string that never appears changes nothing
    r2 = c.post("/generate", json={"prompt": "hello", "num_tokens": 12,
                                   "stop": [" never "]}).json()
    assert r2["tokens"] == full["tokens"]


def test_stop_strings_stream(client_and_worker):
    import json as _json
    c, _ = client_and_worker
    full = c.post("/generate",
                  json={"prompt": "hello", "num_tokens": 12}).json()
    mid = len(full["text"]) // 2
    stop = full["text"][mid:mid + 2]
    with c.stream("POST", "/generate_stream",
                  json={"prompt": "hello", "num_tokens": 12,
                        "stop": [stop]}) as r:
        lines = [ln for ln in r.iter_lines() if ln]
    done = _json.loads(lines[-1].removeprefix("data: "))
    assert done.get("stopped") is True
    assert stop not in done["text"]
    assert full["text"].startswith(done["text"])


def test_trim_at_stop_unit():
    from distributedllm_amd.serving.http import trim_at_stop

    class Tok:  # 1 char per token
        def decode(self, ids):
            return "".join(chr(i) for i in ids)

    ids = [ord(c) for c in "abcXYdef"]
    keep, text, hit = trim_at_stop(Tok(), ids, ["XY"])
    assert hit and text == "abc" and keep == [ord(c) for c in "abc"]
    keep, text, hit = trim_at_stop(Tok(), ids, ["zz"])
    assert not hit and text == "abcXYdef" and len(keep) == len(ids)


def test_openai_completions_endpoint(client_and_worker):
    """OpenAI-compatible /v1/completions maps onto the same batcher
    path: same greedy tokens as /generate, response in the standard
    shape (choices/usage/finish_reason)."""
    c, _ = client_and_worker
    want = c.post("/generate",
                  json={"prompt": "hello", "num_tokens": 6}).json()
    r = c.post("/v1/completions",
               json={"prompt": "hello", "max_tokens": 6}).json()
    assert r["object"] == "text_completion"
    ch = r["choices"][0]
    assert ch["text"] == want["text"]
    assert ch["finish_reason"] == "length"
    assert r["usage"]["completion_tokens"] == 6
    assert r["usage"]["total_tokens"] == (r["usage"]["prompt_tokens"] + 6)
    # stop string -> finish_reason "stop"
    mid = len(want["text"]) // 2
    stop = want["text"][mid:mid + 2]
    r2 = c.post("/v1/completions",
                json={"prompt": "hello", "max_tokens": 6,
                      "stop": stop}).json()
    assert r2["choices"][0]["finish_reason"] == "stop"
    assert stop not in r2["choices"][0]["text"]
    # list prompt (single) accepted; multi rejected
    r3 = c.post("/v1/completions",
                json={"prompt": ["hello"], "max_tokens": 2})
    assert r3.status_code == 200
    r4 = c.post("/v1/completions",
                json={"prompt": ["a", "b"], "max_tokens": 2})
    assert r4.status_code == 400


def test_http_with_batcher_speculation():
    """serve_http --speculate path: an app over a speculating batcher
    returns exactly what the plain app returns."""
    from fastapi.testclient import TestClient
    f = synthetic.build_model("tiny", seed=0)

    def app_for(spec):
        eng = TorchSliceEngine.from_ggml(f, n_ctx=64, max_batch=2)
        eng.attach_extra(slicer.make_extra_layers(f))
        bat = ContinuousBatcher(eng, spec_ngram=3 if spec else 0,
                                spec_k=8 if spec else 0)
        return build_http_app(bat, Tokenizer(f.vocab))

    app_p, w_p = app_for(False)
    app_s, w_s = app_for(True)
    body = {"prompt": "aa aa aa", "num_tokens": 12}
    with TestClient(app_p) as cp, TestClient(app_s) as cs:
        a = cp.post("/generate", json=body).json()
        b = cs.post("/generate", json=body).json()
    w_p.stop()
    w_s.stop()
    assert a["tokens"] == b["tokens"]


def test_concurrent_speculative_chunked_soak():
    """Concurrency soak: threads hammer /generate over a speculating
    batcher with chunked prefill; every response must be complete and
    per-prompt deterministic (scheduling must never leak into
    outputs)."""
    import random
    import threading
    from fastapi.testclient import TestClient
    f = synthetic.build_model("tiny", seed=0)
    eng = TorchSliceEngine.from_ggml(f, n_ctx=64, max_batch=4)
    eng.attach_extra(slicer.make_extra_layers(f))
    bat = ContinuousBatcher(eng, prefill_chunk=8, spec_ngram=3, spec_k=6)
    app, worker = build_http_app(bat, Tokenizer(f.vocab))
    results, errors = {}, []
    lock = threading.Lock()

    def client_thread(tid):
        rng = random.Random(tid)
        with TestClient(app) as c:
            for i in range(6):
                words = " ".join(rng.choice(["aa", "bb", "cc"])
                                 for _ in range(rng.randrange(1, 8)))
                n = rng.randrange(1, 10)
                r = c.post("/generate", json={"prompt": words,
                                              "num_tokens": n})
                if r.status_code != 200:
                    errors.append((tid, i, r.status_code))
                    return
                toks = tuple(r.json()["tokens"])
                with lock:
                    key = (words, n)
                    if len(toks) != n or results.get(key, toks) != toks:
                        errors.append((tid, i, key, toks))
                        return
                    results[key] = toks

    threads = [threading.Thread(target=client_thread, args=(t,))
               for t in range(6)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=300)
    worker.stop()
    assert not errors, errors[:3]
    assert bat.tokens_out >= bat.steps_run  # spec never reduces tokens
