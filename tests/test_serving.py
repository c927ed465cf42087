"""Continuous batcher: per-request prompts, slot reuse, and exact
agreement with the canonical one-request-at-a-time generation loop
(the TCP client's semantics). Capability the reference does not have —
it serves one request at a time (common.py:94-111)."""
import pytest
import torch

from distributedllm_amd.engine import TorchSliceEngine
from distributedllm_amd.engine.sampler import Sampler
from distributedllm_amd.formats import slicer, synthetic
from distributedllm_amd.serving import ContinuousBatcher


def _engine(max_batch):
    f = synthetic.build_model("tiny", seed=0)
    eng = TorchSliceEngine.from_ggml(f, n_ctx=32, max_batch=max_batch)
    eng.attach_extra(slicer.make_extra_layers(f))
    return eng


def _canonical(prompt, steps, sampler=None, eos_id=None):
    """The llm_client loop: prefill prompt, sample from last logits,
    feed only sampled tokens."""
    eng = _engine(1)
    out = []
    cur, n_past = list(prompt), 0
    for _ in range(steps):
        toks = torch.tensor(cur, dtype=torch.int32)
        pos = torch.arange(n_past, n_past + len(cur), dtype=torch.int32)
        seq = torch.zeros(len(cur), dtype=torch.int32)
        y = eng.forward(eng.embed(toks), pos, seq)
        lg = eng.logits(y[-1:].contiguous(), all_logits=True)
        if sampler is None:
            tid = int(torch.argmax(lg[0]).item())
        else:
            tid = sampler(lg[0].numpy())
        out.append(tid)
        if eos_id is not None and tid == eos_id:
            break
        n_past += len(cur)
        cur = [tid]
    return out


PROMPTS = [[5, 9, 3], [7], [2, 11, 4, 6, 1]]
STEPS = [4, 3, 5]


def test_batched_greedy_matches_canonical():
    """Three different-length requests served concurrently must decode
    exactly what each gets when served alone."""
    bat = ContinuousBatcher(_engine(3))
    reqs = [bat.submit(p, s) for p, s in zip(PROMPTS, STEPS)]
    bat.run_all(max_steps=50)
    for r, p, s in zip(reqs, PROMPTS, STEPS):
        assert r.done and r.out == _canonical(p, s)


def test_slot_reuse_under_pressure():
    """max_slots=2 with 3 requests: the third queues, then reuses a
    finished request's KV slot — and still decodes exactly."""
    bat = ContinuousBatcher(_engine(2), max_slots=2)
    reqs = [bat.submit(p, s) for p, s in zip(PROMPTS, STEPS)]
    assert len(bat.queue) == 3
    bat.step()
    assert len(bat.active) == 2 and len(bat.queue) == 1
    bat.run_all(max_steps=50)
    for r, p, s in zip(reqs, PROMPTS, STEPS):
        assert r.done and r.out == _canonical(p, s)
    assert sorted(bat.free) == [0, 1]


def test_per_request_sampler_parity():
    """A sampled request inside the batch draws the same tokens as the
    canonical loop with an identically seeded Sampler (logits order and
    RNG stream both preserved), while a greedy request rides along."""
    want = _canonical(PROMPTS[0], 4, sampler=Sampler(0.8, 1.1, seed=42))
    bat = ContinuousBatcher(_engine(2))
    r0 = bat.submit(PROMPTS[0], 4, sampler=Sampler(0.8, 1.1, seed=42))
    r1 = bat.submit(PROMPTS[1], 3)
    bat.run_all(max_steps=50)
    assert r0.out == want
    assert r1.out == _canonical(PROMPTS[1], 3)


def test_eos_stops_early():
    greedy = _canonical(PROMPTS[0], 4)
    eos = greedy[1]  # stop on the 2nd generated token
    bat = ContinuousBatcher(_engine(1), eos_id=eos)
    r = bat.submit(PROMPTS[0], 10)
    bat.run_all(max_steps=50)
    assert r.done and r.out == greedy[:2] and r.out[-1] == eos


def test_context_overflow_rejected():
    # rejected at submit() time, BEFORE any KV slot is taken, so one bad
    # request can never kill the decode loop
    bat = ContinuousBatcher(_engine(1))
    with pytest.raises(ValueError):
        bat.submit([1] * 30, 10)  # 40 > n_ctx=32
    with pytest.raises(ValueError):
        bat.submit([], 5)
    with pytest.raises(ValueError):
        bat.submit([1, 2], 0)
    # a valid request right at the boundary is accepted and completes
    bat.submit([1] * 16, 16)  # 32 == n_ctx
    done = bat.run_all(max_steps=64)
    assert len(done) == 1 and len(done[0].out) == 16


def test_batch_generate_cli(tmp_path):
    """End-to-end: the batch_generate command serves several prompts
    concurrently from a model file and reports throughput."""
    import subprocess
    import sys
    p = tmp_path / "tiny.bin"
    synthetic.build_model("tiny", seed=0).save(str(p))
    pf = tmp_path / "prompts.txt"
    pf.write_text("hello\nworld\nagain\n")
    out = subprocess.run(
        [sys.executable, "manager.py", "batch_generate", str(p),
         "--prompts-file", str(pf), "--num-tokens", "5", "--greedy",
         "--ctx", "64", "--slots", "2"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr[-1500:]
    lines = out.stdout.strip().splitlines()
    assert len(lines) == 3 and all("->" in ln for ln in lines)
    assert "3 requests, 15 tokens" in out.stderr


def test_multi_lane_batcher_matches_canonical():
    """2 lanes (weight-sharing clone semantics; on CPU two engines over
    the same GGML bytes): slot s lives on lane s%2, tokens must match
    the one-at-a-time loop for every request."""
    f = synthetic.build_model("tiny", seed=0)
    ex = slicer.make_extra_layers(f)
    lanes = []
    for _ in range(2):
        e = TorchSliceEngine.from_ggml(f, n_ctx=32, max_batch=2)
        e.attach_extra(ex)
        lanes.append(e)
    bat = ContinuousBatcher(lanes[0], engines=lanes)
    assert bat.n_slots == 4
    reqs = [bat.submit(p, s) for p, s in zip(PROMPTS, STEPS)]
    bat.run_all(max_steps=50)
    for r, p, s in zip(reqs, PROMPTS, STEPS):
        assert r.done and r.out == _canonical(p, s)


def test_random_arrivals_match_canonical():
    """Randomized schedule stress: random prompts/lengths submitted in
    waves against a 3-slot batcher must each decode exactly the
    canonical continuation (slot reuse, queueing, mixed admission
    prefills all exercised)."""
    import random
    rng = random.Random(0)
    f = synthetic.build_model("tiny", seed=0)
    ex = slicer.make_extra_layers(f)
    eng = TorchSliceEngine.from_ggml(f, n_ctx=32, max_batch=3)
    eng.attach_extra(ex)
    bat = ContinuousBatcher(eng)
    V = f.hparams.n_vocab
    cases = []
    for wave in range(3):
        for _ in range(rng.randint(1, 3)):
            prompt = [rng.randrange(3, V) for _ in
                      range(rng.randint(1, 6))]
            steps = rng.randint(1, 5)
            cases.append((prompt, steps, bat.submit(prompt, steps)))
        for _ in range(rng.randint(1, 4)):  # interleave decode steps
            bat.step()
    bat.run_all(max_steps=100)
    for prompt, steps, r in cases:
        assert r.done and r.out == _canonical(prompt, steps), \
            (prompt, steps, r.out)


def test_local_perplexity_cli(tmp_path, capsys):
    """perplexity accepts a local GGML file (no cluster): same NLL math
    as the distributed client on the same bytes."""
    from distributedllm_amd.cli import execute_command
    p = tmp_path / "tiny.bin"
    synthetic.build_model("tiny", seed=0).save(str(p))
    assert execute_command(["perplexity", str(p),
                            "--prompt", "hello world of words"]) == 0
    out = capsys.readouterr().out
    ppl = float(out.split("perplexity:")[1].strip())
    assert ppl > 1.0


def test_cancel_queued_and_active():
    bat = ContinuousBatcher(_engine(1))
    r1 = bat.submit(PROMPTS[0], 10)
    r2 = bat.submit(PROMPTS[1], 10)   # queued behind r1 (1 slot)
    bat.step()
    assert r1.slot >= 0 and len(bat.queue) == 1
    assert bat.cancel(r2) and r2.done  # cancel while queued
    assert bat.cancel(r1) and r1.done  # cancel while active, frees slot
    assert bat.free == [0] and bat.pending == 0
    # a new request reuses the freed slot and decodes correctly
    r3 = bat.submit(PROMPTS[2], STEPS[2])
    bat.run_all(max_steps=50)
    assert r3.out == _canonical(PROMPTS[2], STEPS[2])
    assert not bat.cancel(r3)  # already finished


def test_chunked_prefill_interleaves_and_stays_exact():
    """prefill_chunk bounds prompt tokens per lane per step: a long
    prompt admits over several steps while an in-flight request keeps
    decoding — and both requests still decode their canonical tokens."""
    eng = _engine(2)
    bat = ContinuousBatcher(eng, prefill_chunk=2)
    long_prompt = [2, 11, 4, 6, 1, 7, 9]  # body of 6 -> 3 chunked steps
    r_short = bat.submit([7], 6)
    bat.step()  # r_short active and decoding
    assert len(r_short.out) == 1
    r_long = bat.submit(long_prompt, 3)
    produced = []
    while not r_long.done:
        bat.step()
        produced.append(len(r_short.out))
    # the short request kept producing tokens during the long prefill
    assert produced[0] > 1 and len(r_short.out) >= 4
    bat.run_all(max_steps=30)
    assert r_short.out == _canonical([7], 6)
    assert r_long.out == _canonical(long_prompt, 3)


def test_speculative_batcher_matches_plain():
    """spec_ngram/spec_k on: per-request outputs must be TOKEN-EXACT
    with the non-speculative batcher (greedy), across mixed prompt
    lengths — including repetitive prompts where drafts accept."""
    prompts = [[7, 7, 7, 7], [5, 9, 3], [4, 8, 2, 4, 8, 2, 4, 8]]
    steps = [12, 8, 10]
    plain = ContinuousBatcher(_engine(3))
    p_reqs = [plain.submit(p, s) for p, s in zip(prompts, steps)]
    plain.run_all(max_steps=100)
    spec = ContinuousBatcher(_engine(3), spec_ngram=2, spec_k=4)
    s_reqs = [spec.submit(p, s) for p, s in zip(prompts, steps)]
    n_steps = 0
    while spec.pending:
        spec.step()
        n_steps += 1
        assert n_steps < 100
    for pr, sr in zip(p_reqs, s_reqs):
        assert sr.done and sr.out == pr.out
    # the repetitive request must have ridden accepted drafts: the
    # whole batch finishes in fewer steps than the longest request's
    # token count would need without speculation
    assert n_steps < max(steps)


def test_speculative_batcher_mixed_sampled():
    """Sampled requests ride along one token per step and stay exact
    (same seeded sampler => same logits sequence => same tokens)."""
    from distributedllm_amd.engine.sampler import Sampler
    prompts = [[7, 7, 7, 7], [5, 9, 3]]
    plain = ContinuousBatcher(_engine(2))
    pa = plain.submit(prompts[0], 10)
    pb = plain.submit(prompts[1], 6, sampler=Sampler(0.8, 1.1, seed=3))
    plain.run_all(max_steps=50)
    spec = ContinuousBatcher(_engine(2), spec_ngram=2, spec_k=4)
    sa = spec.submit(prompts[0], 10)
    sb = spec.submit(prompts[1], 6, sampler=Sampler(0.8, 1.1, seed=3))
    spec.run_all(max_steps=50)
    assert sa.out == pa.out
    assert sb.out == pb.out


def test_speculative_batcher_eos_and_slot_reuse():
    """EOS hit inside an accepted draft ends the request mid-emit and
    frees the slot for the queue."""
    plain = ContinuousBatcher(_engine(1))
    full = plain.submit([7, 7, 7, 7], 16)
    plain.run_all(max_steps=50)
    if len(set(full.out)) < 2:
        return  # degenerate continuation; nothing to cut on
    eos = full.out[len(full.out) // 2]
    want = full.out[:full.out.index(eos) + 1]
    spec = ContinuousBatcher(_engine(1), spec_ngram=2, spec_k=4)
    r1 = spec.submit([7, 7, 7, 7], 16, eos_id=eos)
    r2 = spec.submit([5, 9, 3], 3)   # queued until r1's slot frees
    spec.run_all(max_steps=60)
    assert r1.out == want
    assert r2.done and len(r2.out) == 3


@pytest.mark.parametrize("seed", [0, 1, 2])
def test_speculative_batcher_fuzz_matches_plain(seed):
    """Randomized request mix under slot pressure (queueing + slot
    reuse): speculation must stay token-exact per request."""
    import random
    rng = random.Random(seed)
    jobs = []
    for _ in range(7):
        plen = rng.randrange(1, 7)
        prompt = [rng.randrange(3, 30) for _ in range(plen)]
        if rng.random() < 0.5:           # bias toward repetition
            prompt = prompt[:2] * 3
        jobs.append((prompt, rng.randrange(2, 9)))

    def run(spec):
        # chunked-prefill setting drawn independently per run: outputs
        # must not depend on scheduling at all
        bat = ContinuousBatcher(_engine(2), max_slots=2,
                                spec_ngram=2 if spec else 0,
                                spec_k=4 if spec else 0,
                                prefill_chunk=rng.choice([None, 4]))
        reqs = [bat.submit(p, s) for p, s in jobs]
        bat.run_all(max_steps=500)
        return [r.out for r in reqs]

    assert run(False) == run(True)


def test_batch_generate_speculative_needs_greedy(tmp_path, capsys):
    from distributedllm_amd.cli import execute_command
    f = synthetic.build_model("tiny", seed=0)
    path = tmp_path / "m.bin"
    f.save(str(path))
    assert execute_command(["batch_generate", str(path), "--prompt", "a",
                            "--speculative"]) == 2
