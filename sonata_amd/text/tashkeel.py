"""Arabic diacritization (tashkeel restoration).

Parity: the reference vendors `libtashkeel` (deps/libtashkeel, itself an
ONNX char-level model run through ort; applied before phonemization when
`espeak.voice == "ar"` — piper/src/lib.rs:63-77,251-281).  Here it is a
small char-level neural model (embedding -> dilated conv stack -> softmax
over harakat) run in PyTorch on CPU.  `TashkeelModel.default()` builds a
deterministic random-init instance (no network for real weights; the
checkpoint format is safetensors like every other model here).
"""

from __future__ import annotations

import os
from typing import List, Optional

import torch
from torch import nn

# diacritic classes: none + 8 harakat
_DIACRITICS = ["", "َ", "ُ", "ِ", "ّ", "ْ",
               "ً", "ٌ", "ٍ"]
_AR_MIN, _AR_MAX = 0x0600, 0x06FF
_MAX_LEN = 315  # reference: libtashkeel input cap (~315 chars)


def _char_id(ch: str) -> int:
    cp = ord(ch)
    if _AR_MIN <= cp <= _AR_MAX:
        return cp - _AR_MIN + 2
    return 1  # OOV


class TashkeelNet(nn.Module):
    def __init__(self, vocab: int = 0x100 + 2, emb: int = 64, hidden: int = 128,
                 n_layers: int = 3):
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
        self.head = nn.Conv1d(hidden, len(_DIACRITICS), 1)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        x = self.emb(ids).transpose(1, 2)
        x = self.convs(x)
        return self.head(x)  # [B, n_classes, T]


class TashkeelModel:
    def __init__(self, net: TashkeelNet):
        self.net = net.eval()

    @staticmethod
    def default(device: str = "cpu") -> "TashkeelModel":
        torch.manual_seed(1234)
        net = TashkeelNet().to(device)
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
    def diacritize(self, text: str) -> str:
        """Insert predicted harakat after each Arabic letter.  Text already
        containing diacritics is returned unchanged for those positions."""
        if not text:
            return text
        out: List[str] = []
        for chunk_start in range(0, len(text), _MAX_LEN):
            chunk = text[chunk_start : chunk_start + _MAX_LEN]
            ids = torch.tensor([[_char_id(c) for c in chunk]], dtype=torch.long)
            logits = self.net(ids)[0]  # [n_classes, T]
            pred = logits.argmax(dim=0).tolist()
            for i, ch in enumerate(chunk):
                out.append(ch)
                cp = ord(ch)
                is_letter = _AR_MIN <= cp <= _AR_MAX and not (
                    0x064B <= cp <= 0x0652
                )
                nxt = chunk[i + 1] if i + 1 < len(chunk) else ""
                already = nxt and 0x064B <= ord(nxt) <= 0x0652
                if is_letter and not already:
                    out.append(_DIACRITICS[pred[i]])
        return "".join(out)
