"""Provisioning: prepare a model, slice it per nodes_map, push to nodes.

Parity with the reference pipeline (/root/reference/distllm/cli_api/
provision.py: convert → quantize → extra layers → slices → registry →
push), idempotent at every step (existing outputs are skipped). Sources:

* a GGML/GGJT-v3 model file (f32/f16/q4) — requantized with the in-repo
  q4 codec when the config asks for q4_0/q4_1 (N4 of SURVEY §2.2),
* ``synthetic:<preset>`` — random-init model of a real architecture
  (no-network benchmarking),
* an HF LLaMA directory — converted via formats.hf_convert (N5).

Config (reference README.md:114-133):
    {"model_id": ..., "location": ..., "nodes_map": {"host:port": [a, b]},
     "quantization": "q4_0", "metadata": {...}}
"""
from __future__ import annotations

import json
import os
from typing import Dict

from ..formats import ggml, slicer
from ..formats.synthetic import build_model
from .client import Connection, parse_address
from .registry import ModelEntry, Registry, SliceEntry

VALID_QUANT = {"f32": ggml.FTYPE_ALL_F32, "f16": ggml.FTYPE_MOSTLY_F16,
               "q4_0": ggml.FTYPE_MOSTLY_Q4_0,
               "q4_1": ggml.FTYPE_MOSTLY_Q4_1,
               "q5_0": ggml.FTYPE_MOSTLY_Q5_0,
               "q5_1": ggml.FTYPE_MOSTLY_Q5_1,
               "q8_0": ggml.FTYPE_MOSTLY_Q8_0,
               # k-quant super-block formats (256-weight blocks);
               # requires n_embd/n_ff multiples of 256
               "q2_K": ggml.FTYPE_MOSTLY_Q2_K,
               "q3_K": ggml.FTYPE_MOSTLY_Q3_K_M,
               "q4_K": ggml.FTYPE_MOSTLY_Q4_K_M,
               "q5_K": ggml.FTYPE_MOSTLY_Q5_K_M,
               "q6_K": ggml.FTYPE_MOSTLY_Q6_K}
VALID_FAMILY = {"llama_v1", "llama_v2"}


def validate_metadata(meta: dict) -> None:
    fam = meta.get("family")
    if fam is not None and fam not in VALID_FAMILY:
        raise ValueError(f"unknown model family {fam!r}")
    for key in ("name", "size", "usage_class"):
        v = meta.get(key)
        if v is not None and not str(v).replace("_", "").replace("-", "") \
                .replace(".", "").isalnum():
            raise ValueError(f"metadata field {key}={v!r} is not a safe id")


# upstream-style fallback for tensors whose row length is not a
# multiple of QK_K=256 under a k-quant target (e.g. OpenLLaMA-3B
# E=3200): sub-5-bit targets fall back to q5_0, the rest to q8_0
_KQUANT_FALLBACK = {
    ggml.GGML_TYPE_Q2_K: ggml.GGML_TYPE_Q5_0,
    ggml.GGML_TYPE_Q3_K: ggml.GGML_TYPE_Q5_0,
    ggml.GGML_TYPE_Q4_K: ggml.GGML_TYPE_Q5_0,
    ggml.GGML_TYPE_Q5_K: ggml.GGML_TYPE_Q5_0,
    ggml.GGML_TYPE_Q6_K: ggml.GGML_TYPE_Q8_0,
}


def requantize(f: ggml.GGMLFile, ftype: int) -> ggml.GGMLFile:
    """f16/f32 -> quantized (or dtype change) with the in-repo codecs;
    1-D tensors stay f32 as in real checkpoints."""
    target = ggml._FTYPE_TO_GGML[ftype]
    tensors = []
    for t in f.tensors:
        if len(t.ne) == 1 or t.gtype == target:
            tensors.append(t)
        else:
            gt = target
            if gt in _KQUANT_FALLBACK and t.ne[0] % 256 != 0:
                gt = _KQUANT_FALLBACK[gt]
            tensors.append(
                ggml.GGMLTensor.from_f32(t.name, t.to_f32(), gt))
    hp = f.hparams
    new_hp = ggml.Hparams(hp.n_vocab, hp.n_embd, hp.n_mult, hp.n_head,
                          hp.n_layer, hp.n_rot, ftype, hp.first_layer,
                          n_head_kv=hp.n_head_kv)
    return ggml.GGMLFile(hparams=new_hp, vocab=list(f.vocab),
                         tensors=tensors)


def prepare_model(location: str, quant: str, model_dir: str) -> str:
    """Materialize the (possibly requantized) base GGML model file in
    model_dir; returns its path. Idempotent."""
    os.makedirs(model_dir, exist_ok=True)
    ftype = VALID_QUANT[quant]
    out = os.path.join(model_dir, f"model_{quant}.bin")
    if os.path.exists(out):
        return out
    if location.startswith("synthetic:"):
        preset = location.split(":", 1)[1]
        build_model(preset, ftype=ftype).save(out)
        return out
    if os.path.isdir(location):
        from ..formats.hf_convert import convert_hf_dir
        f = convert_hf_dir(location)
        if f.hparams.ftype != ftype:
            f = requantize(f, ftype)
        f.save(out)
        return out
    f = ggml.GGMLFile.load(location, extended=ggml.sniff_extended(location))
    if f.hparams.ftype != ftype:
        f = requantize(f, ftype)
    f.save(out)
    return out


def provision(config_path: str, root: str = ".", push: bool = True,
              progress=None) -> ModelEntry:
    with open(config_path) as f:
        cfg = json.load(f)
    model_id = cfg["model_id"]
    location = cfg["location"]
    nodes_map: Dict[str, list] = cfg["nodes_map"]
    quant = cfg.get("quantization", "q4_0")
    meta = cfg.get("metadata", {})
    if quant not in VALID_QUANT:
        raise ValueError(f"quantization must be one of {set(VALID_QUANT)}")
    validate_metadata(meta)

    model_dir = os.path.join(root, "models", model_id)
    base_path = prepare_model(location, quant, model_dir)
    base = ggml.GGMLFile.load(base_path, extended=False)
    n_layer = base.hparams.n_layer

    # validate the partition covers [0, n_layer) contiguously
    ranges = sorted(nodes_map.values())
    expect = 0
    for a, b in ranges:
        if a != expect or b < a:
            raise ValueError(
                f"nodes_map ranges must tile layers contiguously; got "
                f"{ranges} for n_layer={n_layer}")
        expect = b + 1
    if expect != n_layer:
        raise ValueError(
            f"nodes_map covers layers [0, {expect}) but model has "
            f"{n_layer}")

    extra_path = os.path.join(model_dir, "extra_layers.bin")
    if not os.path.exists(extra_path):
        slicer.make_extra_layers(base).save(extra_path)

    entry = ModelEntry(model_id=model_id, metadata=meta,
                       model_dir=model_dir, extra_layers_file=extra_path)
    for addr, (a, b) in nodes_map.items():
        spath = os.path.join(model_dir, f"slice_{a}_{b}.bin")
        if not os.path.exists(spath):
            slicer.make_slice(base, a, b).save(spath)
        entry.slices.append(SliceEntry(path=spath, a=a, b=b, address=addr))

    Registry(root).add(entry)

    if push:
        for s in entry.slices:
            host, port = parse_address(s.address)
            conn = Connection(host, port)
            name = os.path.basename(s.path)
            existing = {e["name"] for e in conn.list_slices()}
            if name not in existing:
                conn.push_slice(s.path,
                                metadata={**meta, "name": name,
                                          "model": model_id,
                                          "a": s.a, "b": s.b,
                                          "format": "ggml"},
                                progress=progress)
            conn.close()
    return entry
