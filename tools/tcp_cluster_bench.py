#!/usr/bin/env python3
"""BASELINE.json config 1: N compute-node slices on localhost, the
reference's own workflow (provision -> generate_text through TCP nodes).

Starts N `run_node` processes, provisions a synthetic model sliced
across them, then times greedy generation end-to-end (per-token TCP
round-trips through every node included — the reference pipeline is
fully serialized, one token in flight). On a CPU-only host this is the
reference-path functional config; on a GPU box each node picks up the
HIP engine automatically.

Usage: python tools/tcp_cluster_bench.py [--nodes 2] [--model
open_llama_3b] [--quant q4_0] [--tokens 16] [--workdir DIR]
"""
import argparse
import json
import os
import signal
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def wait_port(port: int, timeout: float = 120.0) -> None:
    import socket
    t0 = time.time()
    while time.time() - t0 < timeout:
        try:
            with socket.create_connection(("127.0.0.1", port), 1.0):
                return
        except OSError:
            time.sleep(0.5)
    raise TimeoutError(f"node on port {port} never came up")


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=2)
    ap.add_argument("--model", default="open_llama_3b")
    ap.add_argument("--quant", default="q4_0")
    ap.add_argument("--tokens", type=int, default=16)
    ap.add_argument("--speculate", type=int, default=0,
                    help="also time a speculative run (greedy, K drafts "
                         "per hop); synthetic random-init continuations "
                         "repeat unrealistically often, so treat the "
                         "acceptance as a mechanism demo, not a real-"
                         "text estimate")
    ap.add_argument("--base-port", type=int, default=19870)
    ap.add_argument("--workdir", default="/tmp/tcp_cluster_bench")
    args = ap.parse_args()

    from distributedllm_amd.models.llama import PRESETS
    n_layer = PRESETS[args.model].n_layer
    per = (n_layer + args.nodes - 1) // args.nodes
    nodes_map, first = {}, 0
    for i in range(args.nodes):
        last = min(n_layer - 1, first + per - 1)
        nodes_map[f"127.0.0.1:{args.base_port + i}"] = [first, last]
        first = last + 1

    os.makedirs(args.workdir, exist_ok=True)
    cfgp = os.path.join(args.workdir, "config.json")
    with open(cfgp, "w") as fh:
        json.dump({"model_id": "bench", "location":
                   f"synthetic:{args.model}",
                   "nodes_map": nodes_map, "quantization": args.quant,
                   "metadata": {"name": "bench", "family": "llama_v1"}},
                  fh)

    procs = []
    try:
        for i in range(args.nodes):
            d = os.path.join(args.workdir, f"node{i}")
            os.makedirs(d, exist_ok=True)
            procs.append(subprocess.Popen(
                [sys.executable, os.path.join(REPO, "manager.py"),
                 "run_node", "--host", "127.0.0.1", "--port",
                 str(args.base_port + i), "--uploads_dir",
                 os.path.join(d, "uploads")],
                stdout=open(os.path.join(d, "log"), "w"),
                stderr=subprocess.STDOUT, cwd=REPO))
        for i in range(args.nodes):
            wait_port(args.base_port + i)

        t0 = time.time()
        subprocess.run([sys.executable, os.path.join(REPO, "manager.py"),
                        "provision", cfgp, "--root", args.workdir],
                       check=True, cwd=REPO, timeout=3600)
        t_prov = time.time() - t0

        def gen():
            t0 = time.time()
            out = subprocess.run(
                [sys.executable, os.path.join(REPO, "manager.py"),
                 "generate_text", cfgp, "--prompt", "Once upon a time",
                 "--num-tokens", str(args.tokens), "--greedy",
                 "--root", args.workdir],
                check=True, cwd=REPO, capture_output=True, text=True,
                timeout=3600)
            return time.time() - t0, out

        def gen_spec():
            t0 = time.time()
            out = subprocess.run(
                [sys.executable, os.path.join(REPO, "manager.py"),
                 "generate_text", cfgp, "--prompt", "Once upon a time",
                 "--num-tokens", str(args.tokens), "--greedy",
                 "--speculate", str(args.speculate),
                 "--root", args.workdir],
                check=True, cwd=REPO, capture_output=True, text=True,
                timeout=3600)
            return time.time() - t0, out

        # cold: includes each node dequantizing + loading its slice on
        # first use; warm: slices stay resident in the node processes
        t_cold, out = gen()
        t_warm, out = gen()
        t_spec = None
        if args.speculate:
            t_spec, out_s = gen_spec()
            plain_txt = out.stdout.splitlines()[0]
            spec_txt = out_s.stdout.splitlines()[0]
            assert spec_txt == plain_txt, "speculative output diverged!"
        print(out.stdout.strip()[-400:])
        print(json.dumps({
            "config": f"{args.model} {args.quant}, {args.nodes} "
                      "compute-node slices on localhost TCP",
            "provision_s": round(t_prov, 1),
            "cold_generate_s": round(t_cold, 2),
            "warm_generate_s": round(t_warm, 2),
            "tokens": args.tokens,
            "tok_s_warm": round(args.tokens / t_warm, 2),
            "spec_generate_s": round(t_spec, 2) if t_spec else None,
            "tok_s_spec": round(args.tokens / t_spec, 2) if t_spec
            else None,
            "note": "end-to-end incl. per-token TCP round-trips through "
                    "every node (reference-workflow serialization)"}),
            flush=True)
        return 0
    finally:
        for p in procs:
            p.send_signal(signal.SIGTERM)
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()


if __name__ == "__main__":
    sys.exit(main())
