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
    return name


def test_name_mapping_roundtrip():
    for ours in ["enc_p.attn_layers.0.conv_q.weight",
                 "enc_p.norm1.2.gamma",
                 "enc_p.ffn_layers.1.conv1.bias",
                 "dp.flows.3.pre.weight",
                 "flow.flows.0.enc.in_layers.2.weight",
                 "dec.resblocks.5.convs1.1.bias",
                 "emb_g.weight"]:
        assert map_vits_name(_to_upstream(ours)) == ours


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
