"""Produce the shipped tashkeel OOV-net weights (deterministic):
    python tools/tools_train_tashkeel.py
writes sonata_amd/text/data/tashkeel.safetensors."""
import sys

sys.path.insert(0, '.')
from sonata_amd.text.tashkeel import (TashkeelModel, train_on_lexicon,
                                      _WEIGHTS)

net = train_on_lexicon(log=True)
TashkeelModel(net).save(_WEIGHTS)
print(f"wrote {_WEIGHTS}")
