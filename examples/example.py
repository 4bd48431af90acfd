"""Usage example (counterpart of the reference python frontend's
example.py): load a voice, synthesize to a file, stream chunks,
inspect RTF.  Works on CPU or MI355X (auto-detected)."""
import sys
import tempfile

sys.path.insert(0, ".")

from sonata_amd.frontends import pysonata
from sonata_amd.models import create_random_voice

# A random-init voice pack (offline environment).  With a real Piper
# voice: python -m sonata_amd.models.onnx_import voice.onnx   and point
# PiperModel at the matching .json config.
pack = create_random_voice(tempfile.mkdtemp(), "example", quality="medium")

model = pysonata.PiperModel(pack)
model.length_scale = 1.0

tts = pysonata.Sonata.with_piper(model)
print("language:", tts.language, "| sample rate:",
      tts.get_audio_output_info().sample_rate)

# one-shot to file
tts.synthesize_to_file("example.wav", "Hello world. This is sonata on MI355X.")

# per-sentence results with timing
for wave in tts.synthesize("Hello again. How are you today?"):
    print(f"sentence: {wave.duration_ms:.0f} ms audio, "
          f"inference {wave.inference_ms:.1f} ms, RTF {wave.real_time_factor:.4f}")

# realtime chunk stream (bytes)
n = sum(len(c) for c in tts.synthesize_streamed(
    "Streaming synthesis delivers audio while the sentence still decodes."))
print(f"streamed {n} PCM bytes")

# multilingual: every supported language code works the same way
# (full inventory in docs/LANGUAGES.md)
for lang, text in [("hi", "नमस्ते दुनिया।"), ("ko", "안녕하세요 세계."),
                   ("ja", "こんにちは、世界。"), ("cmn", "你好，世界。"),
                   ("yue", "你好，世界。")]:
    p = create_random_voice(tempfile.mkdtemp(), f"ex_{lang}",
                            quality="x_low", language=lang)
    t = pysonata.Sonata.with_piper(pysonata.PiperModel(p))
    print(lang, "->", pysonata.phonemize_text(text, language=lang))
    t.synthesize_to_file(f"example_{lang}.wav", text)
