"""CLI surface + proxy relay + HF conversion, end-to-end on CPU.

Covers the reference capabilities of cli_api/ (all 9 commands),
proxy_node.py (reverse-connect relay), and vendored convert.py
(HF → GGML), per SURVEY §1 L6/L7 and §2.2 N5.
"""
import json
import os
import threading
import time

import numpy as np
import pytest

from distributedllm_amd.cli import build_parser, execute_command
from distributedllm_amd.cluster.client import Connection
from distributedllm_amd.cluster.node import NodeServer, NodeState
from distributedllm_amd.cluster.proxy import ProxyServer, connect_then_serve
from distributedllm_amd.formats import ggml
from distributedllm_amd.formats.synthetic import build_model
from distributedllm_amd.models.llama import PRESETS


@pytest.fixture()
def node(tmp_path):
    srv = NodeServer("127.0.0.1", 0, str(tmp_path / "uploads"))
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield srv
    srv.shutdown()
    srv.server_close()


def _write_dummy(tmp_path, k=2.0, b=1.0):
    path = tmp_path / "dummy.bin"
    np.array([k, b], dtype="<f4").tofile(path)
    return str(path)


def test_all_commands_registered():
    from distributedllm_amd.cli.base import commands
    # the reference's 9 commands plus the batched-serving extension
    assert set(commands) == {
        "provision", "run_node", "status", "push_slice", "load_slice",
        "list_slices", "generate_text", "perplexity", "run_proxy",
        "batch_generate", "serve_http"}
    build_parser()  # parser builds without error


def test_cli_push_load_status_list(node, tmp_path, capsys):
    addr = f"127.0.0.1:{node.port}"
    dummy = _write_dummy(tmp_path)
    assert execute_command(["push_slice", addr, dummy,
                            "--metadata", '{"format": "test"}']) == 0
    assert execute_command(["list_slices", addr]) == 0
    out = capsys.readouterr().out
    assert "dummy.bin" in out
    assert execute_command(["load_slice", addr, "dummy.bin"]) == 0
    assert execute_command(["status", addr]) == 0
    out = capsys.readouterr().out
    assert '"model": "dummy.bin"' in out


def test_cli_provision_generate_perplexity(node, tmp_path, capsys,
                                           monkeypatch):
    """Full reference workflow via the CLI: provision a synthetic tiny
    model to one node, generate text, compute perplexity."""
    addr = f"127.0.0.1:{node.port}"
    root = tmp_path / "root"
    root.mkdir()
    cfg = {"model_id": "tiny_test",
           "location": "synthetic:tiny",
           "nodes_map": {addr: [0, PRESETS["tiny"].n_layer - 1]},
           "quantization": "f16",
           "metadata": {"name": "tiny", "family": "llama_v1",
                        "size": "3B", "usage_class": "test"}}
    cfg_path = tmp_path / "cfg.json"
    cfg_path.write_text(json.dumps(cfg))

    assert execute_command(["provision", str(cfg_path),
                            "--root", str(root)]) == 0
    assert execute_command(["generate_text", str(cfg_path),
                            "--prompt", "hello world", "--num-tokens", "3",
                            "--greedy", "--root", str(root)]) == 0
    out = capsys.readouterr().out
    assert "hello world" in out

    assert execute_command(["perplexity", str(cfg_path),
                            "--prompt", "hello world of words",
                            "--root", str(root)]) == 0
    out = capsys.readouterr().out
    assert "perplexity:" in out
    ppl = float(out.split("perplexity:")[1].strip())
    assert np.isfinite(ppl) and ppl > 1.0

    # provisioning is idempotent: second run reuses everything
    assert execute_command(["provision", str(cfg_path),
                            "--root", str(root)]) == 0


def test_perplexity_requires_one_source(tmp_path):
    assert execute_command(["perplexity", "nonexistent.json"]) == 2


def test_proxy_reverse_node_roundtrip(tmp_path):
    """Client → proxy → reverse-connected node → back, through the real
    framing (reference proxy_node.py capability)."""
    proxy = ProxyServer("127.0.0.1", 0, 0)
    proxy.start()
    state = NodeState(str(tmp_path / "uploads"))
    stop = threading.Event()
    t = threading.Thread(target=connect_then_serve,
                         args=("127.0.0.1", proxy.node_port, state),
                         kwargs={"stop": stop}, daemon=True)
    t.start()

    # wait for the reverse connection to land
    deadline = time.time() + 5.0
    conn = Connection("127.0.0.1", proxy.client_port)
    while time.time() < deadline:
        try:
            if conn.get_status().status == "up":
                break
        except Exception:
            time.sleep(0.05)
    else:
        pytest.fail("reverse node never reachable through proxy")

    dummy = _write_dummy(tmp_path, k=3.0, b=-1.0)
    conn.push_slice(dummy, metadata={"format": "test"})
    conn.load_slice("dummy.bin")
    x = np.arange(6, dtype=np.float32).reshape(2, 3)
    y = conn.propagate_forward(x, start_pos=0)
    np.testing.assert_allclose(y, 3.0 * x - 1.0)
    conn.close()
    stop.set()
    proxy.shutdown()


def test_proxy_no_node_errors():
    from distributedllm_amd.cluster.client import OperationFailedError
    proxy = ProxyServer("127.0.0.1", 0, 0)
    proxy.start()
    conn = Connection("127.0.0.1", proxy.client_port)
    with pytest.raises(OperationFailedError, match="no_node"):
        conn.get_status()
    conn.close()
    proxy.shutdown()


# ------------------------------------------------------------- hf_convert

def _fake_hf_dir(tmp_path, base: ggml.GGMLFile):
    """Build an HF-layout dir whose weights are the inverse-permuted
    tensors of `base`, so convert_hf_dir must reproduce `base` exactly."""
    from distributedllm_amd.formats.hf_convert import (_LAYER_MAP, _TOP_MAP,
                                                       permute_rotary)
    hp = base.hparams
    d = tmp_path / "hf"
    d.mkdir()
    (d / "config.json").write_text(json.dumps({
        "hidden_size": hp.n_embd, "num_attention_heads": hp.n_head,
        "num_hidden_layers": hp.n_layer, "vocab_size": hp.n_vocab,
        "intermediate_size": hp.n_ff}))

    def unpermute(w, n_head):
        rows = w.shape[0]
        return (w.reshape(n_head, rows // n_head // 2, 2, *w.shape[1:])
                 .swapaxes(1, 2).reshape(w.shape))

    # sanity: unpermute inverts permute
    probe = np.arange(hp.n_embd * 4, dtype=np.float32).reshape(hp.n_embd, 4)
    np.testing.assert_array_equal(
        unpermute(permute_rotary(probe, hp.n_head), hp.n_head), probe)

    inv_layer = {v: k for k, v in _LAYER_MAP.items()}
    inv_top = {v: k for k, v in _TOP_MAP.items()}
    sd = {}
    for t in base.tensors:
        a = t.to_f32()
        if t.name in inv_top:
            sd[inv_top[t.name]] = a
            continue
        _, idx, suffix = t.name.split(".", 2)
        if suffix in ("attention.wq.weight", "attention.wk.weight"):
            a = unpermute(a, hp.n_head)
        sd[f"model.layers.{idx}.{inv_layer[suffix]}"] = a

    from safetensors.numpy import save_file
    save_file(sd, str(d / "model.safetensors"))
    return str(d)


def test_hf_convert_roundtrip(tmp_path):
    base = build_model("tiny", ftype=ggml.FTYPE_MOSTLY_F16, seed=3)
    hf_dir = _fake_hf_dir(tmp_path, base)
    from distributedllm_amd.formats.hf_convert import convert_hf_dir
    conv = convert_hf_dir(hf_dir)
    hp, chp = base.hparams, conv.hparams
    # n_mult may differ (any value reproducing n_ff is valid) — n_ff must match
    assert (chp.n_vocab, chp.n_embd, chp.n_ff, chp.n_head, chp.n_layer,
            chp.n_rot) == (hp.n_vocab, hp.n_embd, hp.n_ff, hp.n_head,
                           hp.n_layer, hp.n_rot)
    bm, cm = base.tensor_map(), conv.tensor_map()
    assert set(bm) == set(cm)
    for name in bm:
        np.testing.assert_allclose(cm[name].to_f32(), bm[name].to_f32(),
                                   atol=1e-3, rtol=1e-3,
                                   err_msg=name)


def test_find_n_mult_real_models():
    from distributedllm_amd.formats.hf_convert import find_n_mult
    # (E, n_ff) pairs of the real checkpoints (SURVEY §2.5 n_ff formula)
    for name in ("open_llama_3b", "llama_7b", "llama_13b", "llama_30b",
                 "llama_65b"):
        p = PRESETS[name]
        m = find_n_mult(p.n_embd, p.n_ff)
        assert ((2 * (4 * p.n_embd) // 3 + m - 1) // m) * m == p.n_ff


def test_control_center(node, tmp_path):
    """Cluster status model (reference ControlCenter capability)."""
    from distributedllm_amd.cluster.control import ControlCenter, ModelSlice
    addr = f"127.0.0.1:{node.port}"
    cc = ControlCenter({addr: [0, 2]})
    st = cc.get_status()[addr]
    assert st.connectivity and not st.slice_loaded
    ready, why = cc.pipeline_ready(n_layer=3)
    assert not ready and "no slice" in why

    cc.validate_slices(4, [ModelSlice("a", 0, 1), ModelSlice("b", 2, 3)])
    with pytest.raises(ValueError):
        cc.validate_slices(4, [ModelSlice("a", 0, 1), ModelSlice("b", 3, 3)])

    # unreachable node
    cc2 = ControlCenter({"127.0.0.1:1": [0, 0]})
    assert not cc2.get_status()["127.0.0.1:1"].connectivity


def test_perplexity_dataset_source(node, tmp_path, capsys):
    """--dataset samples a random prompt from a local HF dataset
    (reference perplexity.py:35-51 capability, offline)."""
    import datasets
    d = datasets.Dataset.from_dict(
        {"text": ["hello world of words and more words", "another text"]})
    ds_path = tmp_path / "ds"
    d.save_to_disk(str(ds_path))

    addr = f"127.0.0.1:{node.port}"
    root = tmp_path / "root"; root.mkdir()
    cfg = {"model_id": "tiny_ds", "location": "synthetic:tiny",
           "nodes_map": {addr: [0, PRESETS["tiny"].n_layer - 1]},
           "quantization": "f16", "metadata": {"name": "tinyds"}}
    cfg_path = tmp_path / "cfg.json"
    cfg_path.write_text(json.dumps(cfg))
    assert execute_command(["provision", str(cfg_path),
                            "--root", str(root)]) == 0
    assert execute_command(["perplexity", str(cfg_path),
                            "--dataset", str(ds_path), "--seed", "0",
                            "--root", str(root)]) == 0
    assert "perplexity:" in capsys.readouterr().out


def test_status_config_cluster(node, tmp_path, capsys):
    addr = f"127.0.0.1:{node.port}"
    cfg = {"model_id": "x", "location": "synthetic:tiny",
           "nodes_map": {addr: [0, 2]}}
    cfg_path = tmp_path / "c.json"
    cfg_path.write_text(json.dumps(cfg))
    # nodes up but no slice loaded -> not ready (exit 1 with --n-layer)
    assert execute_command(["status", "--config", str(cfg_path),
                            "--n-layer", "3"]) == 1
    out = capsys.readouterr().out
    assert '"connectivity": true' in out and "pipeline_ready: False" in out


def test_cli_provision_q8_0_generate(node, tmp_path, capsys):
    """Provisioning with a classic non-q4 quantization (q8_0) runs the
    whole requantize -> slice -> push -> load -> generate flow (the
    reference engine accepts q5/q8 GGJT files; so do our nodes)."""
    addr = f"127.0.0.1:{node.port}"
    root = tmp_path / "rootq8"
    root.mkdir()
    cfg = {"model_id": "tiny_q8",
           "location": "synthetic:tiny",
           "nodes_map": {addr: [0, PRESETS["tiny"].n_layer - 1]},
           "quantization": "q8_0",
           "metadata": {"name": "tinyq8", "family": "llama_v1"}}
    cfg_path = tmp_path / "cfgq8.json"
    cfg_path.write_text(json.dumps(cfg))
    assert execute_command(["provision", str(cfg_path),
                            "--root", str(root)]) == 0
    from distributedllm_amd.formats import ggml
    base = ggml.GGMLFile.load(
        str(root / "models" / "tiny_q8" / "model_q8_0.bin"), extended=False)
    assert base.hparams.ftype == ggml.FTYPE_MOSTLY_Q8_0
    assert execute_command(["generate_text", str(cfg_path),
                            "--prompt", "hi", "--num-tokens", "2",
                            "--greedy", "--root", str(root)]) == 0
    assert "hi" in capsys.readouterr().out


def test_cli_clean_error_on_unreachable_node(tmp_path, capsys):
    """Operational failures (node down) print one error line, exit 1 —
    no traceback."""
    cfg = {"model_id": "x", "location": "synthetic:tiny",
           "nodes_map": {"127.0.0.1:1": [0, PRESETS["tiny"].n_layer - 1]},
           "quantization": "f16", "metadata": {"name": "x"}}
    p = tmp_path / "cfg.json"
    p.write_text(json.dumps(cfg))
    rc = execute_command(["provision", str(p), "--root", str(tmp_path)])
    assert rc == 1
    err = capsys.readouterr().err
    assert "error:" in err and "Traceback" not in err


def test_tcp_cluster_bench_tool(tmp_path):
    """tools/tcp_cluster_bench.py (BASELINE config 1 driver) runs the
    whole provision -> generate flow against real local nodes."""
    import json as _json
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "tools",
                                      "tcp_cluster_bench.py"),
         "--nodes", "2", "--model", "tiny", "--tokens", "4",
         "--base-port", str(21870 + os.getpid() % 500),
         "--workdir", str(tmp_path)],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines()
            if l.startswith("{") and "tok_s_warm" in l][-1]
    r = _json.loads(line)
    assert r["tokens"] == 4 and r["tok_s_warm"] > 0


def test_tcp_generate_speculative_exact(node, tmp_path):
    """Speculative generation over the TCP pipeline must emit exactly
    the plain-greedy continuation (drafts verified per hop; rejections
    rewind the client's n_past against the stateless node)."""
    from distributedllm_amd.cluster.llm_client import get_llm
    addr = f"127.0.0.1:{node.port}"
    root = tmp_path / "root"
    root.mkdir()
    cfg = {"model_id": "tiny_spec",
           "location": "synthetic:tiny",
           "nodes_map": {addr: [0, PRESETS["tiny"].n_layer - 1]},
           "quantization": "f16",
           "metadata": {"name": "tiny", "family": "llama_v1"}}
    cfg_path = tmp_path / "cfg.json"
    cfg_path.write_text(json.dumps(cfg))
    assert execute_command(["provision", str(cfg_path),
                            "--root", str(root)]) == 0
    llm = get_llm(str(cfg_path), root=str(root))
    for prompt in ("aaaa aaaa aaaa", "hello"):
        plain = list(llm.generate(prompt, max_steps=16, greedy=True))
        spec = list(llm.generate(prompt, max_steps=16, greedy=True,
                                 speculative=6))
        assert spec == plain, prompt
