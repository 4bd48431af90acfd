"""C ABI frontend tests: drive libsonata_amd.so through ctypes exactly as
a C caller would (callback protocol, config round-trip, speak-to-file) —
mirroring the reference C API semantics (crates/frontends/capi)."""

import ctypes as C
import os

import pytest

LIB = os.path.join(os.path.dirname(__file__), "..", "sonata_amd",
                   "frontends", "libsonata_amd.so")

pytestmark = pytest.mark.skipif(
    not os.path.exists(LIB), reason="libsonata_amd.so not built")


class ExternError(C.Structure):
    _fields_ = [("code", C.c_int32), ("message", C.c_char_p)]


class SynthesisEvent(C.Structure):
    _fields_ = [("event_type", C.c_int32),
                ("error_ptr", C.POINTER(ExternError)),
                ("len", C.c_int64),
                ("data", C.POINTER(C.c_uint8))]


class AudioInfo(C.Structure):
    _fields_ = [("sample_rate", C.c_uint32), ("num_channels", C.c_uint32),
                ("sample_width", C.c_uint32)]


class PiperSynthConfig(C.Structure):
    _fields_ = [("speaker", C.c_uint32), ("length_scale", C.c_float),
                ("noise_scale", C.c_float), ("noise_w", C.c_float)]


CALLBACK = C.CFUNCTYPE(C.c_uint8, SynthesisEvent)


class SynthesisParams(C.Structure):
    _fields_ = [("mode", C.c_int32), ("rate", C.c_uint8),
                ("volume", C.c_uint8), ("pitch", C.c_uint8),
                ("appended_silence_ms", C.c_uint32),
                ("callback", CALLBACK), ("nonblocking", C.c_uint8)]


@pytest.fixture(scope="module")
def lib():
    lib = C.CDLL(LIB)
    lib.libsonataLoadVoiceFromConfigPath.restype = C.c_void_p
    lib.libsonataLoadVoiceFromConfigPath.argtypes = [
        C.c_char_p, C.POINTER(ExternError)]
    lib.libsonataUnloadSonataVoice.argtypes = [C.c_void_p]
    lib.libsonataGetAudioInfo.argtypes = [
        C.c_void_p, C.POINTER(AudioInfo), C.POINTER(ExternError)]
    lib.libsonataGetPiperDefaultSynthConfig.restype = \
        C.POINTER(PiperSynthConfig)
    lib.libsonataGetPiperDefaultSynthConfig.argtypes = [
        C.c_void_p, C.POINTER(ExternError)]
    lib.libsonataSetPiperSynthConfig.argtypes = [
        C.c_void_p, PiperSynthConfig, C.POINTER(ExternError)]
    lib.libsonataSpeak.argtypes = [
        C.c_void_p, C.c_char_p, SynthesisParams, C.POINTER(ExternError)]
    lib.libsonataSpeakToFile.restype = C.c_uint8
    lib.libsonataSpeakToFile.argtypes = [
        C.c_void_p, C.c_char_p, SynthesisParams, C.c_char_p,
        C.POINTER(ExternError)]
    lib.libsonataFreePiperSynthConfig.argtypes = [
        C.POINTER(PiperSynthConfig)]
    return lib


@pytest.fixture(scope="module")
def voice(lib, tmp_path_factory):
    from sonata_amd.models import create_random_voice

    d = tmp_path_factory.mktemp("capi_voice")
    pack = create_random_voice(str(d), "capi_voice", quality="x_low")
    err = ExternError()
    os.environ["SONATA_DEVICE"] = "cpu"
    h = lib.libsonataLoadVoiceFromConfigPath(pack.encode(), C.byref(err))
    assert err.code == 0, err.message
    assert h
    yield h
    lib.libsonataUnloadSonataVoice(h)


def test_load_bad_path(lib):
    err = ExternError()
    h = lib.libsonataLoadVoiceFromConfigPath(b"/nope/voice.json",
                                             C.byref(err))
    assert not h
    assert err.code == 17  # FAILED_TO_LOAD_RESOURCE


def test_audio_info(lib, voice):
    err, info = ExternError(), AudioInfo()
    lib.libsonataGetAudioInfo(voice, C.byref(info), C.byref(err))
    assert err.code == 0
    assert info.sample_rate == 16000
    assert info.num_channels == 1 and info.sample_width == 2


def test_synth_config_roundtrip(lib, voice):
    err = ExternError()
    cfg_p = lib.libsonataGetPiperDefaultSynthConfig(voice, C.byref(err))
    assert err.code == 0
    cfg = cfg_p.contents
    cfg.length_scale = 1.25
    lib.libsonataSetPiperSynthConfig(voice, cfg, C.byref(err))
    assert err.code == 0
    back = lib.libsonataGetPiperDefaultSynthConfig(voice, C.byref(err))
    assert abs(back.contents.length_scale - 1.25) < 1e-6
    lib.libsonataFreePiperSynthConfig(cfg_p)
    lib.libsonataFreePiperSynthConfig(back)


def test_speak_callback_protocol(lib, voice):
    events = []

    @CALLBACK
    def cb(ev):
        if ev.event_type == 0:  # SPEECH
            events.append(bytes(C.cast(
                ev.data, C.POINTER(C.c_uint8 * ev.len)).contents))
        else:
            events.append(ev.event_type)
        return 0

    err = ExternError()
    params = SynthesisParams(mode=1, rate=0, volume=0, pitch=0,
                             appended_silence_ms=0, callback=cb,
                             nonblocking=0)
    lib.libsonataSpeak(voice, "hˈɛloʊ wˈɜːld.".encode(), params,
                       C.byref(err))
    assert err.code == 0
    assert events[-1] == 1  # FINISHED
    chunks = [e for e in events if isinstance(e, bytes)]
    assert len(chunks) >= 1 and sum(len(c) for c in chunks) > 500


def test_speak_invalid_mode(lib, voice):
    err = ExternError()
    params = SynthesisParams(mode=9, callback=CALLBACK(lambda ev: 0),
                             nonblocking=0)
    lib.libsonataSpeak(voice, b"x.", params, C.byref(err))
    assert err.code == 16  # INVALID_SYNTHESIS_MODE


def test_speak_to_file(lib, voice, tmp_path):
    err = ExternError()
    out = str(tmp_path / "c.wav").encode()
    params = SynthesisParams(mode=1, callback=CALLBACK(lambda ev: 0),
                             nonblocking=0)
    ok = lib.libsonataSpeakToFile(voice, "tˈɛst sˈɛntəns.".encode(),
                                  params, out, C.byref(err))
    assert ok == 1 and err.code == 0
    assert open(out, "rb").read(4) == b"RIFF"


def test_speak_nonblocking(lib, voice):
    """Nonblocking mode: libsonataSpeak returns immediately; events arrive
    on a worker thread (reference capi lib.rs:366-386)."""
    import threading
    import time

    done = threading.Event()
    chunks = []

    @CALLBACK
    def cb(ev):
        if ev.event_type == 0:
            chunks.append(int(ev.len))
        elif ev.event_type == 1:
            done.set()
        return 0

    err = ExternError()
    params = SynthesisParams(mode=1, callback=cb, nonblocking=1)
    t0 = time.perf_counter()
    lib.libsonataSpeak(voice, "hˈɛloʊ wˈɜːld ˈɛvɹiwˌʌn.".encode(), params,
                       C.byref(err))
    returned_in = time.perf_counter() - t0
    assert err.code == 0
    assert done.wait(timeout=120), "FINISHED event never arrived"
    assert sum(chunks) > 500
    assert returned_in < 5.0  # returned before synthesis completed


def test_native_engine_engaged(lib, voice):
    """The C library must run on the native C++ engine (GIL-free
    synthesis hot path), not the Python fallback."""
    lib.libsonataIsNativeEngine.restype = C.c_uint8
    lib.libsonataIsNativeEngine.argtypes = [C.c_void_p]
    assert lib.libsonataIsNativeEngine(voice) == 1


def test_native_matches_python_path(lib, voice, tmp_path):
    """Native C-API synthesis == the Python engine path for the same
    text (same per-utterance seed derivation via the shared bridge)."""
    import numpy as np

    # reset the module-shared voice to default scales (an earlier test
    # may have changed them); the python comparison uses defaults
    err0 = ExternError()
    cfg_p = lib.libsonataGetPiperDefaultSynthConfig(voice, C.byref(err0))
    cfg = cfg_p.contents
    cfg.length_scale, cfg.noise_scale, cfg.noise_w = 1.0, 0.667, 0.8
    cfg.speaker = 0
    lib.libsonataSetPiperSynthConfig(voice, cfg, C.byref(err0))
    lib.libsonataFreePiperSynthConfig(cfg_p)

    chunks = []

    @CALLBACK
    def cb(ev):
        if ev.event_type == 0 and ev.len:
            chunks.append(bytes(C.cast(
                ev.data, C.POINTER(C.c_uint8 * ev.len)).contents))
        return 0

    params = SynthesisParams(mode=0, rate=0, volume=0, pitch=0,
                             appended_silence_ms=0, callback=cb,
                             nonblocking=0)
    err = ExternError()
    lib.libsonataSpeak(voice, "One two three.".encode(), params,
                       C.byref(err))
    assert err.code == 0 and chunks
    native = np.frombuffer(b"".join(chunks), dtype=np.int16)

    # same text through the pure-Python path (CPU voice, same pack)
    from sonata_amd.audio.samples import to_i16
    from sonata_amd.models.voice import load_voice
    import glob

    packs = glob.glob(os.path.join(
        os.path.dirname(str(tmp_path)), "capi_voice*", "*.json"))
    v = load_voice(packs[0], device="cpu", engine="python")
    phon = v.phonemize_text("One two three.").sentences
    py = np.concatenate(
        [to_i16(v.speak_one_sentence(p).samples) for p in phon])
    assert len(native) == len(py), (len(native), len(py))
    # identical seeds + same kernels (CPU f32) -> near-identical PCM
    assert float(np.abs(native.astype(np.int32)
                        - py.astype(np.int32)).max()) <= 16


def test_native_realtime_mode_streams(lib, voice):
    counts = []

    @CALLBACK
    def cb(ev):
        if ev.event_type == 0:
            counts.append(ev.len)
        return 0

    params = SynthesisParams(mode=2, rate=0, volume=0, pitch=0,
                             appended_silence_ms=0, callback=cb,
                             nonblocking=0)
    err = ExternError()
    lib.libsonataSpeak(
        voice,
        "This is a much longer sentence that should stream in several "
        "chunks of audio rather than one.".encode(),
        params, C.byref(err))
    assert err.code == 0
    assert len(counts) >= 2, counts  # actually chunked
    assert all(c > 0 for c in counts)


def test_native_concurrent_voices(lib, tmp_path_factory):
    """Two voices synthesize concurrently from two C threads (ctypes
    releases the GIL during the call; the native path never takes it
    for the graph).  Correctness + no deadlock."""
    import threading

    from sonata_amd.models import create_random_voice

    os.environ["SONATA_DEVICE"] = "cpu"
    handles = []
    for i in range(2):
        d = tmp_path_factory.mktemp(f"capi_cc{i}")
        pack = create_random_voice(str(d), f"cc{i}", quality="x_low",
                                   seed=i)
        err = ExternError()
        h = lib.libsonataLoadVoiceFromConfigPath(pack.encode(),
                                                 C.byref(err))
        assert err.code == 0 and h
        handles.append(h)

    results = [None, None]
    cbs = []

    def make_cb(idx):
        got = []

        @CALLBACK
        def cb(ev):
            if ev.event_type == 0 and ev.len:
                got.append(ev.len)
            elif ev.event_type == 1:
                results[idx] = sum(got)
            return 0

        cbs.append(cb)  # keep alive
        return cb

    threads = []
    for i, h in enumerate(handles):
        params = SynthesisParams(mode=0, rate=0, volume=0, pitch=0,
                                 appended_silence_ms=0,
                                 callback=make_cb(i), nonblocking=0)

        def run(hh=h, pp=params):
            err = ExternError()
            lib.libsonataSpeak(hh, b"Concurrent synthesis test.", pp,
                               C.byref(err))

        threads.append(threading.Thread(target=run))
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
    for i, h in enumerate(handles):
        assert results[i] and results[i] > 1000, results
        lib.libsonataUnloadSonataVoice(h)


@pytest.mark.gpu
def test_capi_native_on_gpu(lib, tmp_path_factory):
    """Native C-API synthesis on the GPU engine (bf16 kernels), driven
    through ctypes exactly as a C caller would."""
    import numpy as np

    from sonata_amd.models import create_random_voice

    d = tmp_path_factory.mktemp("capi_gpu")
    pack = create_random_voice(str(d), "gpuv", quality="medium")
    os.environ["SONATA_DEVICE"] = "cuda:0"
    try:
        err = ExternError()
        h = lib.libsonataLoadVoiceFromConfigPath(pack.encode(),
                                                 C.byref(err))
        assert err.code == 0 and h
        lib.libsonataIsNativeEngine.restype = C.c_uint8
        lib.libsonataIsNativeEngine.argtypes = [C.c_void_p]
        assert lib.libsonataIsNativeEngine(h) == 1

        chunks = []

        @CALLBACK
        def cb(ev):
            if ev.event_type == 0 and ev.len:
                chunks.append(bytes(C.cast(
                    ev.data, C.POINTER(C.c_uint8 * ev.len)).contents))
            return 0

        for mode in (0, 2):  # lazy one-shot + realtime streaming
            chunks.clear()
            params = SynthesisParams(mode=mode, rate=0, volume=0, pitch=0,
                                     appended_silence_ms=0, callback=cb,
                                     nonblocking=0)
            err = ExternError()
            lib.libsonataSpeak(
                h, "One two three four five six seven.".encode(),
                params, C.byref(err))
            assert err.code == 0, err.message
            pcm = np.frombuffer(b"".join(chunks), dtype=np.int16)
            assert len(pcm) > 10000
            assert np.abs(pcm).max() > 1000  # real audio, not zeros
        # to-file path
        out = str(d / "gpu.wav")
        params = SynthesisParams(mode=0, rate=0, volume=0, pitch=0,
                                 appended_silence_ms=0,
                                 callback=C.cast(None, CALLBACK),
                                 nonblocking=0)
        ok = lib.libsonataSpeakToFile(h, b"Writing a file.", params,
                                      out.encode(), C.byref(err))
        assert ok == 1 and os.path.getsize(out) > 1000
        lib.libsonataUnloadSonataVoice(h)
    finally:
        os.environ["SONATA_DEVICE"] = "cpu"
