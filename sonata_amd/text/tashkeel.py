"""Arabic diacritization (tashkeel restoration).

Parity: the reference vendors `libtashkeel` (deps/libtashkeel, itself an
ONNX char-level model run through ort; applied before phonemization when
`espeak.voice == "ar"` — piper/src/lib.rs:63-77,251-281).

This implementation is a two-tier diacritizer that actually works offline:
  1. LEXICON tier (tashkeel_lexicon.py): ~260 high-frequency stems with
     full clitic morphology (wa-/fa-/bi-/li-/ka-/al- prefixes, possessive
     suffixes, sun-letter assimilation) — high precision on the function
     words and common vocabulary that dominate real token streams.
  2. NEURAL tier: char-level conv net for out-of-lexicon words.  The
     shipped weights (data/tashkeel.safetensors) are TRAINED on the forms
     the lexicon machinery expands (train_on_lexicon) — it generalizes
     common orthographic patterns (al- sukun, CV templates); it is NOT
     the reference's libtashkeel model (no network for those weights).
     `import_tashkeel_onnx` is the compatibility path for a real model
     export whose parameter tree matches; unknown layouts fail loudly.
"""

from __future__ import annotations

import os
import re
from typing import Dict, List, Optional

import torch
from torch import nn

from .tashkeel_lexicon import lookup, strip_diacritics

FATHA, DAMMA, KASRA, SUKUN, SHADDA = "َ", "ُ", "ِ", "ْ", "ّ"
# diacritic classes predicted after each consonant (shadda precedes its
# vowel in the codepoint stream)
_DIACRITICS = ["", FATHA, DAMMA, KASRA, "ً", "ٌ", "ٍ", SUKUN, SHADDA,
               SHADDA + FATHA, SHADDA + DAMMA, SHADDA + KASRA,
               SHADDA + "ً"]
_DIA_TO_ID = {d: i for i, d in enumerate(_DIACRITICS)}
_DIA_CHARS = set("ًٌٍَُِّْ")
_AR_MIN, _AR_MAX = 0x0600, 0x06FF
_MAX_LEN = 315  # reference: libtashkeel input cap (~315 chars)
_WORD_RE = re.compile(r"[ء-يٱ-ۓ]+")
_WEIGHTS = os.path.join(os.path.dirname(__file__), "data",
                        "tashkeel.safetensors")


def _char_id(ch: str) -> int:
    cp = ord(ch)
    if _AR_MIN <= cp <= _AR_MAX:
        return cp - _AR_MIN + 2
    return 1  # OOV


class TashkeelNet(nn.Module):
    def __init__(self, vocab: int = 0x100 + 2, emb: int = 64,
                 hidden: int = 128, n_layers: int = 3,
                 n_classes: int = len(_DIACRITICS)):
        super().__init__()
        self.emb = nn.Embedding(vocab, emb)
        convs = []
        ch = emb
        for i in range(n_layers):
            convs.append(nn.Conv1d(ch, hidden, 3, padding=3 ** i,
                                   dilation=3 ** i))
            convs.append(nn.ReLU())
            ch = hidden
        self.convs = nn.Sequential(*convs)
        self.head = nn.Conv1d(hidden, n_classes, 1)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        x = self.emb(ids).transpose(1, 2)
        x = self.convs(x)
        return self.head(x)  # [B, n_classes, T]


def _split_diacritized(form: str):
    """Diacritized string -> (bare chars, per-char diacritic class)."""
    bare: List[str] = []
    classes: List[int] = []
    for ch in form:
        if ch in _DIA_CHARS and bare:
            cur = _DIACRITICS[classes[-1]] + ch
            classes[-1] = _DIA_TO_ID.get(cur, classes[-1])
        else:
            bare.append(ch)
            classes.append(0)
    return "".join(bare), classes


def train_on_lexicon(epochs: int = 300, seed: int = 7,
                     log: bool = False) -> "TashkeelNet":
    """Train the OOV net on every form the lexicon machinery expands.
    Deterministic; used to produce the shipped data/tashkeel.safetensors
    (tools/tools_train_tashkeel.py)."""
    from .tashkeel_lexicon import expand_training_forms

    torch.manual_seed(seed)
    net = TashkeelNet()
    forms = expand_training_forms()
    pairs = [_split_diacritized(f) for f in forms]
    T = max(len(b) for b, _ in pairs)
    ids = torch.ones(len(pairs), T, dtype=torch.long) * 0
    tgt = torch.full((len(pairs), T), -100, dtype=torch.long)
    for i, (bare, classes) in enumerate(pairs):
        ids[i, : len(bare)] = torch.tensor(
            [_char_id(c) for c in bare], dtype=torch.long)
        tgt[i, : len(classes)] = torch.tensor(classes, dtype=torch.long)
    opt = torch.optim.Adam(net.parameters(), lr=3e-3)
    loss_fn = nn.CrossEntropyLoss(ignore_index=-100)
    net.train()
    for ep in range(epochs):
        opt.zero_grad()
        logits = net(ids)
        loss = loss_fn(logits, tgt)
        loss.backward()
        opt.step()
        if log and ep % 50 == 0:
            acc = ((logits.argmax(1) == tgt) & (tgt >= 0)).sum() / (
                tgt >= 0).sum()
            print(f"epoch {ep}: loss {loss.item():.4f} acc {acc:.3f}")
    net.eval()
    return net


class TashkeelModel:
    def __init__(self, net: TashkeelNet):
        self.net = net.eval()

    @staticmethod
    def default(device: str = "cpu") -> "TashkeelModel":
        """Shipped trained weights when present, else train-on-import
        (deterministic, a few seconds on CPU)."""
        net = TashkeelNet()
        if os.path.exists(_WEIGHTS):
            from safetensors.torch import load_file

            net.load_state_dict(load_file(_WEIGHTS))
            net = net.to(device)
        else:  # pragma: no cover - shipped weights exist in the repo
            net = train_on_lexicon().to(device)
        return TashkeelModel(net)

    @staticmethod
    def load(path: str, device: str = "cpu") -> "TashkeelModel":
        from safetensors.torch import load_file

        net = TashkeelNet().to(device)
        net.load_state_dict(load_file(path))
        return TashkeelModel(net)

    def save(self, path: str) -> None:
        from safetensors.torch import save_file

        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        save_file(self.net.state_dict(), path)

    @torch.no_grad()
    def _net_diacritize_word(self, word: str) -> str:
        ids = torch.tensor([[_char_id(c) for c in word]], dtype=torch.long)
        pred = self.net(ids)[0].argmax(dim=0).tolist()
        out = []
        for ch, cls in zip(word, pred):
            out.append(ch + _DIACRITICS[cls])
        return "".join(out)

    @torch.no_grad()
    def diacritize(self, text: str) -> str:
        """Lexicon/clitic lookup per word; neural net for OOV words.
        Words already carrying any diacritic are passed through verbatim
        (the author's tashkeel wins)."""
        if not text:
            return text

        def repl(m: re.Match) -> str:
            word = m.group(0)
            if any(c in _DIA_CHARS for c in word):
                return word
            hit = lookup(word)
            if hit is not None:
                return hit
            if len(word) <= 1:
                return word
            return self._net_diacritize_word(word[:_MAX_LEN]) \
                + word[_MAX_LEN:]

        return _WORD_RE.sub(repl, text)


# --------------------------------------------------------------------- #
# ONNX weight importer (compatibility path for a real tashkeel export)
# --------------------------------------------------------------------- #
_ONNX_RULES = [
    (r"^(embedding|emb)\.weight$", "emb.weight"),
    (r"^convs?\.(\d+)\.(weight|bias)$", r"convs.\1.\2"),
    (r"^(head|proj|classifier|output)\.(weight|bias)$", r"head.\2"),
]


def import_tashkeel_onnx(onnx_path: str, out_path: str = None,
                         allow_partial: bool = False) -> str:
    """Convert a tashkeel ONNX model into our safetensors checkpoint.

    Parses the protobuf initializers (shared reader with the voice
    importer), maps names through _ONNX_RULES, infers (vocab, emb,
    hidden, n_layers, n_classes) from the mapped shapes, and verifies
    FULL coverage of the TashkeelNet tree — a layout this importer does
    not understand (e.g. recurrent nets) fails loudly instead of
    producing a half-random diacritizer (mirrors onnx_import.py)."""
    import numpy as np

    from ..core import ModelError
    from ..models.onnx_import import parse_onnx_initializers

    inits = parse_onnx_initializers(onnx_path)
    if not inits:
        raise ModelError(f"no initializers found in {onnx_path}")
    state: Dict[str, torch.Tensor] = {}
    skipped: List[str] = []
    for name, arr in inits.items():
        mapped = None
        for pat, repl in _ONNX_RULES:
            if re.match(pat, name):
                mapped = re.sub(pat, repl, name)
                break
        if mapped is None or not np.issubdtype(arr.dtype, np.floating):
            skipped.append(name)
            continue
        state[mapped] = torch.from_numpy(
            np.ascontiguousarray(arr.astype(np.float32)))
    if "emb.weight" not in state or not any(
            k.startswith("head.") for k in state):
        raise ModelError(
            f"tashkeel onnx layout not recognized: mapped keys "
            f"{sorted(state)[:6]}, skipped {skipped[:6]} — extend "
            f"_ONNX_RULES for this export")
    vocab, emb = state["emb.weight"].shape
    conv_ids = sorted({int(k.split(".")[1]) for k in state
                       if k.startswith("convs.")})
    n_layers = len(conv_ids)
    # our Sequential interleaves ReLU (indices 0,2,4..); a plain export
    # may number convs densely (0,1,2..) — renumber to our slots
    remap = {old: 2 * i for i, old in enumerate(conv_ids)}
    renumbered: Dict[str, torch.Tensor] = {}
    for k, v in state.items():
        if k.startswith("convs."):
            parts = k.split(".")
            parts[1] = str(remap[int(parts[1])])
            renumbered[".".join(parts)] = v
        else:
            renumbered[k] = v
    state = renumbered
    hidden = state[f"convs.{remap[conv_ids[-1]]}.weight"].shape[0]
    n_classes = state["head.weight"].shape[0]
    net = TashkeelNet(vocab=vocab, emb=emb, hidden=hidden,
                      n_layers=n_layers, n_classes=n_classes)
    want = set(net.state_dict().keys())
    have = set(state.keys())
    missing = sorted(want - have)
    if missing and not allow_partial:
        raise ModelError(
            f"tashkeel import does not cover the net: missing "
            f"{missing[:6]} ({len(missing)}); skipped {skipped[:6]}")
    net.load_state_dict({k: v for k, v in state.items() if k in want},
                        strict=not allow_partial)
    if out_path is None:
        stem = onnx_path
        if stem.endswith(".onnx"):
            stem = stem[: -len(".onnx")]
        out_path = stem + ".safetensors"
    from safetensors.torch import save_file

    save_file(net.state_dict(), out_path)
    return out_path
