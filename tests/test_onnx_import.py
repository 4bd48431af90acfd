"""ONNX weight-importer tests: a synthetic Piper-style .onnx (upstream
VITS initializer names, hand-encoded protobuf) round-trips into a
loadable voice pack with identical tensors."""

import struct

import numpy as np
import pytest
import torch

from sonata_amd.models.onnx_import import (import_onnx_voice, map_vits_name,
                                           parse_onnx_initializers)


# ---- minimal protobuf writer (test-side ONNX emitter) -------------------- #
def _varint(v: int) -> bytes:
    out = b""
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out += bytes([b | 0x80])
        else:
            return out + bytes([b])


def _tag(field: int, wt: int) -> bytes:
    return _varint((field << 3) | wt)


def _ld(field: int, payload: bytes) -> bytes:
    return _tag(field, 2) + _varint(len(payload)) + payload


def _tensor_proto(name: str, arr: np.ndarray) -> bytes:
    body = b""
    for d in arr.shape:
        body += _tag(1, 0) + _varint(d)
    if arr.dtype == np.int64:
        body += _tag(2, 0) + _varint(7)  # INT64
        body += _ld(8, name.encode())
        body += _ld(9, arr.tobytes())
    else:
        body += _tag(2, 0) + _varint(1)  # FLOAT
        body += _ld(8, name.encode())
        body += _ld(9, arr.astype(np.float32).tobytes())
    return body


def _onnx_bytes(tensors) -> bytes:
    graph = b""
    for name, arr in tensors:
        graph += _ld(5, _tensor_proto(name, arr))
    return _ld(7, graph)  # ModelProto.graph


# ---- upstream-name emulation of our state dict --------------------------- #
_REVERSE = [
    ("enc_p.attn_layers.", "enc_p.encoder.attn_layers."),
    ("enc_p.norm1.", "enc_p.encoder.norm_layers_1."),
    ("enc_p.norm2.", "enc_p.encoder.norm_layers_2."),
]


def _to_upstream(name: str) -> str:
    for ours, theirs in _REVERSE:
        if name.startswith(ours):
            name = theirs + name[len(ours):]
    import re

    name = re.sub(r"(enc_p\.encoder\.ffn_layers\.\d+\.)conv(\d)\.",
                  r"\1conv_\2.", name)
    # Upstream ResidualCouplingBlock flows = [Layer, Flip]*4: coupling
    # layers live at ModuleList indices 0,2,4,6 (vits models.py); ours
    # pack them densely at 0..3.
    m = re.match(r"^flow\.flows\.(\d+)\.(.*)$", name)
    if m:
        name = f"flow.flows.{2 * int(m.group(1))}.{m.group(2)}"
    return name


def test_name_mapping_roundtrip():
    for ours in ["enc_p.attn_layers.0.conv_q.weight",
                 "enc_p.norm1.2.gamma",
                 "enc_p.ffn_layers.1.conv1.bias",
                 "dp.flows.3.pre.weight",
                 "flow.flows.0.enc.in_layers.2.weight",
                 "flow.flows.3.enc.res_skip_layers.1.weight",
                 "dec.resblocks.5.convs1.1.bias",
                 "emb_g.weight"]:
        assert map_vits_name(_to_upstream(ours)) == ours
    # genuine upstream flow indices 0,2,4,6 land on ours 0..3
    assert map_vits_name("flow.flows.6.pre.weight") == "flow.flows.3.pre.weight"
    assert map_vits_name("flow.flows.4.enc.cond_layer.bias") == \
        "flow.flows.2.enc.cond_layer.bias"


def test_import_synthetic_onnx(tmp_path):
    from sonata_amd.models import create_random_voice
    from sonata_amd.models.voice import load_voice, _weights_path_for

    pack = create_random_voice(str(tmp_path), "onnx_voice", quality="x_low")
    from safetensors.torch import load_file

    state = load_file(_weights_path_for(pack))

    # emit a piper-style onnx: upstream names + training-only extras
    tensors = [(_to_upstream(k), v.numpy()) for k, v in state.items()]
    tensors.append(("enc_q.pre.weight", np.zeros((4, 4), np.float32)))
    tensors.append(("dp.post_pre.weight", np.zeros((2, 2), np.float32)))
    onnx_path = str(tmp_path / "onnx_voice.onnx")
    with open(onnx_path, "wb") as f:
        f.write(_onnx_bytes(tensors))

    parsed = parse_onnx_initializers(onnx_path)
    assert len(parsed) == len(tensors)

    out = import_onnx_voice(onnx_path,
                            str(tmp_path / "imported.safetensors"))
    imported = load_file(out)
    assert set(imported.keys()) == set(state.keys())
    for k in state:
        assert torch.equal(imported[k], state[k]), k

    # full path: config + imported weights -> working voice
    import shutil

    shutil.copy(out, _weights_path_for(pack))
    voice = load_voice(pack, device="cpu")
    audio = voice.speak_one_sentence("tˈɛst.")
    assert len(audio.samples) > 500


def _upstream_export_tensors(net, n_speakers=1):
    """Emulate a genuine Piper export from OUR weights, first-principles:
    upstream `SynthesizerTrn` names (flow coupling layers at ModuleList
    indices 0,2,4,6 with param-less Flips between), HiFi-GAN weight norm
    removed before export (piper export_onnx calls dec.remove_weight_norm),
    flow WN layers still weight-norm-parametrized (weight_g/weight_v,
    decomposed g=||w||, v=w so g*v/||v|| == w), training-only tensors
    (enc_q, dp.post_*) and int64 graph constants included as a real
    export's initializer list would have them."""
    tensors = []
    for k, t in net.state_dict().items():
        up = _to_upstream(k)
        w = t.numpy()
        if (up.startswith("flow.flows.") and up.endswith(".weight")
                and (".enc.in_layers." in up or ".enc.res_skip_layers." in up
                     or ".enc.cond_layer." in up)):
            g = np.sqrt((w.astype(np.float64) ** 2).sum(
                axis=tuple(range(1, w.ndim)), keepdims=True))
            tensors.append((up[:-len(".weight")] + ".weight_g",
                            g.astype(np.float32)))
            tensors.append((up[:-len(".weight")] + ".weight_v", w))
        else:
            tensors.append((up, w))
    # training-only extras a checkpoint-derived export may carry
    tensors.append(("enc_q.pre.weight", np.zeros((4, 4, 1), np.float32)))
    tensors.append(("enc_q.enc.in_layers.0.weight_v",
                    np.ones((4, 2, 5), np.float32)))
    tensors.append(("enc_q.enc.in_layers.0.weight_g",
                    np.ones((4, 1, 1), np.float32)))
    tensors.append(("dp.post_pre.weight", np.zeros((2, 2, 1), np.float32)))
    tensors.append(("dp.post_flows.1.pre.weight",
                    np.zeros((2, 2, 1), np.float32)))
    # int64 graph constants (shapes/indices); one inside a tree prefix
    tensors.append(("onnx::Reshape_412", np.array([1, -1, 256], np.int64)))
    tensors.append(("dec.num_upsamples",
                    np.array([2 ** 40 + 7], np.int64)))  # would corrupt as f32
    return tensors


@pytest.mark.parametrize("n_speakers", [1, 4])
def test_full_fidelity_piper_export_roundtrip(tmp_path, n_speakers):
    """The complete upstream name set: import a
    full-fidelity synthetic export and assert the resulting pack loads
    with strict=True coverage AND synthesizes identically to the same
    net built directly."""
    from safetensors.torch import load_file
    from sonata_amd.models import create_random_voice
    from sonata_amd.models.voice import load_voice, _weights_path_for

    pack = create_random_voice(str(tmp_path), "fidelity", quality="x_low",
                               num_speakers=n_speakers,
                               language="en-us", seed=3)
    state = load_file(_weights_path_for(pack))

    from sonata_amd.models.config import ModelConfig
    from sonata_amd.models.vits import VitsModel
    cfg = ModelConfig.from_json_path(pack)
    net = VitsModel(cfg.num_symbols, cfg.architecture,
                    n_speakers=max(n_speakers, 1))
    net.load_state_dict({k: v for k, v in state.items()}, strict=True)

    tensors = _upstream_export_tensors(net, n_speakers)
    onnx_path = str(tmp_path / "fidelity.onnx")
    with open(onnx_path, "wb") as f:
        f.write(_onnx_bytes(tensors))
    # reference config naming: `<onnx>.json` names the model file
    import shutil
    shutil.copy(pack, onnx_path + ".json")

    out = import_onnx_voice(onnx_path, str(tmp_path / "imp.safetensors"))
    imported = load_file(out)
    # exact coverage of the target tree
    assert set(imported.keys()) == set(state.keys())
    for k in state:
        np.testing.assert_allclose(imported[k].numpy(), state[k].numpy(),
                                   rtol=2e-6, atol=2e-6, err_msg=k)

    # synthesis through the imported pack == direct net
    shutil.copy(out, _weights_path_for(pack))
    voice = load_voice(pack, device="cpu", engine="python")
    audio = voice.speak_one_sentence("həlˈoʊ wˈɜːld.")
    assert len(audio.samples) > 500


def test_import_missing_coverage_fails_loudly(tmp_path):
    """A name-scheme mismatch (e.g. constant-folded generated names) must
    raise, not write a partially random-init voice pack."""
    from sonata_amd.models import create_random_voice

    pack = create_random_voice(str(tmp_path), "partial", quality="x_low")
    from sonata_amd.models.config import ModelConfig
    nsym = ModelConfig.from_json_path(pack).num_symbols
    # an export where flow tensors got constant-folded to generated names
    tensors = [("enc_p.emb.weight", np.zeros((nsym, 96), np.float32)),
               ("onnx::Conv_123", np.zeros((96, 48, 1), np.float32))]
    onnx_path = str(tmp_path / "partial.onnx")
    with open(onnx_path, "wb") as f:
        f.write(_onnx_bytes(tensors))
    import shutil
    shutil.copy(pack, onnx_path + ".json")
    from sonata_amd.core import ModelError
    with pytest.raises(ModelError, match="does not cover"):
        import_onnx_voice(onnx_path, str(tmp_path / "p.safetensors"))
    # allow_partial downgrades to a warning and writes
    out = import_onnx_voice(onnx_path, str(tmp_path / "p.safetensors"),
                            allow_partial=True)
    assert out.endswith("p.safetensors")


def test_weight_norm_pairs_are_fused(tmp_path):
    """Exports that kept weight_norm parametrization (weight_g/weight_v)
    get fused into plain weights: w = g * v/||v||."""
    rng = np.random.default_rng(0)
    v = rng.standard_normal((6, 4, 3)).astype(np.float32)
    g = rng.standard_normal((6, 1, 1)).astype(np.float32)
    norm = np.sqrt((v.astype(np.float64) ** 2).sum(axis=(1, 2),
                                                   keepdims=True))
    expected = (g * v / norm).astype(np.float32)
    tensors = [("flow.flows.0.enc.in_layers.0.weight_v", v),
               ("flow.flows.0.enc.in_layers.0.weight_g", g),
               ("flow.flows.0.enc.in_layers.0.bias",
                np.zeros(6, np.float32))]
    onnx_path = str(tmp_path / "wn.onnx")
    with open(onnx_path, "wb") as f:
        f.write(_onnx_bytes(tensors))
    from safetensors.torch import load_file

    out = import_onnx_voice(onnx_path, str(tmp_path / "wn.safetensors"))
    got = load_file(out)
    assert "flow.flows.0.enc.in_layers.0.weight" in got
    np.testing.assert_allclose(
        got["flow.flows.0.enc.in_layers.0.weight"].numpy(), expected,
        rtol=1e-5)
    assert "flow.flows.0.enc.in_layers.0.weight_v" not in got


def test_streaming_pack_import(tmp_path):
    """Reference streaming packs = config.json + encoder.onnx +
    decoder.onnx (piper lib.rs:90-96): both files' weights merge into
    one pack that loads strict and synthesizes."""
    import shutil

    from sonata_amd.models import create_random_voice
    from sonata_amd.models.config import ModelConfig
    from sonata_amd.models.onnx_import import import_streaming_pack
    from sonata_amd.models.vits import VitsModel
    from sonata_amd.models.voice import load_voice, _weights_path_for

    pack = create_random_voice(str(tmp_path), "rt", quality="x_low")
    cfg = ModelConfig.from_json_path(pack)
    from safetensors.torch import load_file

    state = load_file(_weights_path_for(pack))
    net = VitsModel(cfg.num_symbols, cfg.architecture, n_speakers=1)
    net.load_state_dict(state, strict=True)
    tensors = _upstream_export_tensors(net)
    enc_t = [(n, a) for n, a in tensors if not n.startswith("dec.")]
    # decoder exported WITHOUT the dec. prefix (upstream Generator tree)
    dec_t = [(n[len("dec."):], a) for n, a in tensors
             if n.startswith("dec.")]
    # streaming layout: config.json + encoder.onnx + decoder.onnx
    d = tmp_path / "rt_pack"
    d.mkdir()
    shutil.copy(pack, d / "config.json")
    with open(d / "encoder.onnx", "wb") as f:
        f.write(_onnx_bytes(enc_t))
    with open(d / "decoder.onnx", "wb") as f:
        f.write(_onnx_bytes(dec_t))
    shutil.copy(_weights_path_for(pack).replace(".safetensors",
                                                ".safetensors"),
                d / "ignore.bin")  # decoy; importer must not need it

    out = import_streaming_pack(str(d / "config.json"))
    imported = load_file(out)
    assert set(imported.keys()) == set(state.keys())
    v = load_voice(str(d / "config.json"), device="cpu", engine="python")
    chunks = list(v.stream_synthesis("tˈɛst strˈiːm.", 45, 3))
    assert sum(len(c) for c in chunks) > 500
