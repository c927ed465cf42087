"""k-quant (q2_K .. q6_K) super-block codecs + file/engine plumbing.

Dequantization follows the upstream super-block bit layouts exactly;
the quantizers are simple nearest-rounding (upstream runs an
error-minimizing search), so round-trip CONSISTENCY is the asserted
property, plus hand-computed known-vector decodes for the layout bits.
"""
import numpy as np
import pytest

from distributedllm_amd.formats import ggml, kquants, slicer, synthetic

RNG = np.random.default_rng(7)


@pytest.mark.parametrize("name", list(kquants.CODECS))
def test_roundtrip_accuracy_and_fixed_point(name):
    bb, qz, dq = kquants.CODECS[name]
    x = RNG.standard_normal((8, 512)).astype(np.float32) * 0.1
    raw = qz(x)
    assert raw.shape == (8, 2 * bb)
    back = dq(raw, 512)
    rel = np.linalg.norm(back - x) / np.linalg.norm(x)
    # expected reconstruction error per bit width
    limit = {"q2_K": 0.40, "q3_K": 0.25, "q4_K": 0.12, "q5_K": 0.06,
             "q6_K": 0.03}[name]
    assert rel < limit, (name, rel)
    # quantize(dequantize(.)) must (nearly) fix-point
    again = dq(qz(back), 512)
    rel2 = np.linalg.norm(again - back) / np.linalg.norm(back)
    assert rel2 < 0.08, (name, rel2)


def test_q6_k_known_vector():
    """Hand-built super-block: d=1, scales[0]=2 (others 1), quants set
    so the layout positions (ql low/high nibble, qh 2-bit planes) are
    each exercised."""
    raw = np.zeros(kquants.Q6_K_BLOCK_BYTES, dtype=np.uint8)
    raw[208:210] = np.frombuffer(np.float16(1.0).tobytes(), np.uint8)
    sc = np.ones(16, dtype=np.int8)
    sc[0] = 2
    raw[192:208] = sc.view(np.uint8)
    # weight 0 (half 0, k 0, l 0): ql[0] low nibble + qh[0] bits 0-1
    raw[0] = 0x5          # low = 5
    raw[128] = 0x1        # high bits 01 -> q = 5 + 16 = 21
    w = kquants.dequantize_q6_K(raw[None, :], 256)[0]
    assert w[0] == (21 - 32) * 2.0        # scale[0] = 2
    # weight 64 (half 0, k 2): ql[0] HIGH nibble, qh[0] bits 4-5
    raw2 = np.zeros_like(raw)
    raw2[208:210] = raw[208:210]
    raw2[192:208] = np.ones(16, np.int8).view(np.uint8)
    raw2[0] = 0x30        # high nibble = 3
    raw2[128] = 0x10      # bits 4-5 = 01 -> q = 3 + 16 = 19
    w2 = kquants.dequantize_q6_K(raw2[None, :], 256)[0]
    assert w2[64] == (19 - 32) * 1.0
    assert w2[0] == (0 - 32) * 1.0


def test_q4_k_scale_packing_roundtrip():
    sc = RNG.integers(0, 64, (5, 8)).astype(np.uint8)
    mn = RNG.integers(0, 64, (5, 8)).astype(np.uint8)
    p = kquants._pack_scales_k4(sc, mn)
    sc2, mn2 = kquants._unpack_scales_k4(p)
    assert np.array_equal(sc, sc2) and np.array_equal(mn, mn2)


def test_q3_k_scale_packing_roundtrip():
    sc = RNG.integers(-32, 32, (5, 16)).astype(np.int16)
    p = kquants._pack_scales_q3(sc)
    sc2 = kquants._unpack_scales_q3(p)
    assert np.array_equal(sc, sc2.astype(np.int16))


@pytest.mark.parametrize("ftype", [ggml.FTYPE_MOSTLY_Q4_K_M,
                                   ggml.FTYPE_MOSTLY_Q6_K])
def test_kquant_model_roundtrip_and_slice(ftype, tmp_path):
    """A k-quant GGML model saves/loads and slices (tensor sizes and
    alignment honoured end to end); C++ slice_model passes the k-quant
    bytes through identically."""
    import subprocess
    from pathlib import Path
    f = synthetic.build_model("small_k", ftype=ftype, seed=1)
    p = tmp_path / "m.bin"
    f.save(str(p))
    g = ggml.GGMLFile.load(str(p), extended=False)
    t = g.tensor_map()["layers.0.feed_forward.w1.weight"]
    assert t.gtype == ggml._FTYPE_TO_GGML[ftype]
    assert t.to_f32().shape == (1536, 512)
    out_py = tmp_path / "s_py.bin"
    slicer.make_slice(g, 0, 0).save(str(out_py))
    tools = Path(__file__).resolve().parent.parent / "tools" / "bin"
    out_c = tmp_path / "s_c.bin"
    subprocess.run([str(tools / "slice_model"), "slice", str(p), "0", "0",
                    str(out_c)], check=True, capture_output=True)
    assert out_c.read_bytes() == out_py.read_bytes()


def test_requantize_kquant_fallback():
    """Rows not divisible by 256 (OpenLLaMA-3B class) fall back to
    q5_0/q8_0 per tensor, like upstream quantize."""
    from distributedllm_amd.cluster.provision import requantize
    f = synthetic.build_model("tiny", ftype=ggml.FTYPE_MOSTLY_F16, seed=0)
    g = requantize(f, ggml.FTYPE_MOSTLY_Q4_K_M)   # E=64: all fall back
    t = g.tensor_map()["layers.0.attention.wq.weight"]
    assert t.gtype == ggml.GGML_TYPE_Q5_0
    g2 = requantize(f, ggml.FTYPE_MOSTLY_Q6_K)
    assert g2.tensor_map()["layers.0.attention.wq.weight"].gtype == \
        ggml.GGML_TYPE_Q8_0


def test_kquant_byte_expansion_matches_codec():
    """The engine's byte expansion (repack input) must reproduce the
    codec's dequantized values exactly: w = alpha*(u-128) + beta."""
    from distributedllm_amd.engine.slice_engine import (
        _K16_GTYPES, _K32_GTYPES, _kquant_byte_values)
    x = RNG.standard_normal((16, 512)).astype(np.float32) * 0.1
    for gt in _K32_GTYPES + _K16_GTYPES:
        t = ggml.GGMLTensor.from_f32("w", x, gt)
        want = t.to_f32()
        vals, alpha, beta = _kquant_byte_values(t)
        u = vals.astype(np.float32) - 128.0
        if gt in _K32_GTYPES:
            got = alpha[..., None] * u + beta[..., None]
        else:
            u16 = u.reshape(u.shape[0], u.shape[1], 2, 16)
            got = (alpha[..., :, None] * u16 +
                   beta[..., :, None]).reshape(u.shape)
        got = got.reshape(want.shape)
        err = np.abs(got - want).max()
        assert err < 1e-5, (ggml.TYPE_NAMES[gt], err)
