// libsonata_amd C ABI implementation.
//
// Behavior parity with the reference C API (crates/frontends/capi/src/
// lib.rs): opaque voice handles (:40-64), error codes (:19-26), chunked
// callback protocol with SPEECH/FINISHED/ERROR events and nonzero-return
// cancellation (:415-438), nonblocking dispatch on a worker thread
// (:366-386), realtime chunking 72/3 (:407-409).
//
// Architecture (r2): the SYNTHESIS hot path runs through the native C++
// VitsEngine directly — a C caller holds the GIL only for the text
// front-end (phonemize/tashkeel/per-utterance seeds, milliseconds) and
// optional prosody DSP; the whole neural graph + PCM conversion runs
// GIL-free (the previous implementation bracketed
// every synthesis in the interpreter).  If the native engine cannot
// load (e.g. missing weights format), the original full-Python bridge
// path is the fallback.
//
// Threading: Python touchpoints bracket PyGILState_Ensure/Release, so
// the library works from plain C programs (interpreter initialized on
// first use) and inside an existing Python process (ctypes).  Native
// engine calls are serialized per voice handle (the engine's weight
// cache is not thread-safe); different voices synthesize concurrently.
#include "libsonata_amd.h"

#include <Python.h>
#include <torch/torch.h>

#include <cstdlib>
#include <cstring>
#include <fstream>
#include <functional>
#include <mutex>
#include <string>
#include <thread>
#include <utility>
#include <vector>

#include "../engine/vits_engine.h"
#include "audio_util.h"

namespace {

std::once_flag g_init_once;

void ensure_interpreter() {
  std::call_once(g_init_once, [] {
    if (!Py_IsInitialized()) {
      Py_InitializeEx(0);
      // Drop the GIL acquired by Py_Initialize so that any thread can
      // PyGILState_Ensure later.
      PyEval_SaveThread();
    }
  });
}

struct GIL {
  PyGILState_STATE st;
  GIL() { st = PyGILState_Ensure(); }
  ~GIL() { PyGILState_Release(st); }
};

void set_error(ExternError *err, ErrorCode code, const std::string &msg) {
  if (!err) return;
  err->code = code;
  err->message = static_cast<char *>(std::malloc(msg.size() + 1));
  std::memcpy(err->message, msg.c_str(), msg.size() + 1);
}

void set_success(ExternError *err) {
  if (!err) return;
  err->code = ErrorCode_SUCCESS;
  err->message = nullptr;
}

// Fetch the pending Python exception as a string (clears it).
std::string py_error_string() {
  PyObject *type = nullptr, *value = nullptr, *tb = nullptr;
  PyErr_Fetch(&type, &value, &tb);
  std::string out = "unknown python error";
  if (value) {
    PyObject *s = PyObject_Str(value);
    if (s) {
      const char *c = PyUnicode_AsUTF8(s);
      if (c) out = c;
      Py_DECREF(s);
    }
  }
  Py_XDECREF(type);
  Py_XDECREF(value);
  Py_XDECREF(tb);
  return out;
}

PyObject *bridge() {  // borrowed-style: cached module reference
  static PyObject *mod = nullptr;
  if (!mod) mod = PyImport_ImportModule("sonata_amd.frontends.capi_bridge");
  return mod;
}

// Call bridge.<fn>(*args); returns new reference or nullptr.
PyObject *bridge_call(const char *fn, PyObject *args) {
  PyObject *mod = bridge();
  if (!mod) return nullptr;
  PyObject *f = PyObject_GetAttrString(mod, fn);
  if (!f) return nullptr;
  PyObject *r = PyObject_CallObject(f, args);
  Py_DECREF(f);
  return r;
}

ErrorCode classify(const std::string &msg) {
  if (msg.find("Phonemization") != std::string::npos)
    return PHONEMIZATION_ERROR;
  if (msg.find("not found") != std::string::npos ||
      msg.find("No such file") != std::string::npos ||
      msg.find("weights") != std::string::npos)
    return FAILED_TO_LOAD_RESOURCE;
  return OPERATION_ERROR;
}

struct VoiceHandle {
  sonata::VitsEngine *engine = nullptr;  // native synthesis (hot path)
  PyObject *frontend = nullptr;          // phonemize-only bridge object
  PyObject *synth = nullptr;             // full-Python fallback
  std::mutex mu;                         // serializes engine calls
  // native synthesis config (reference PiperSynthConfig)
  unsigned int speaker = 0;
  float length_scale = 1.0f, noise_scale = 0.667f, noise_w = 0.8f;
  long sample_rate = 22050;
  long num_speakers = 1;
};

// ---- native synthesis helpers ----------------------------------------- //

// phonemize + seeds under the GIL; false on failure (python error set)
bool frontend_phonemize(VoiceHandle *h, const std::string &text,
                        unsigned int speaker,
                        std::vector<std::pair<std::string, int64_t>> *out,
                        std::string *err_msg) {
  GIL gil;
  PyObject *args = Py_BuildValue("(OsI)", h->frontend, text.c_str(), speaker);
  PyObject *r = args ? bridge_call("phonemize_with_seeds", args) : nullptr;
  Py_XDECREF(args);
  if (!r) {
    *err_msg = py_error_string();
    return false;
  }
  bool ok = true;
  PyObject *iter = PyObject_GetIter(r);
  PyObject *item;
  while (iter && (item = PyIter_Next(iter))) {
    const char *ph = nullptr;
    long long seed = 0;
    if (PyArg_ParseTuple(item, "sL", &ph, &seed)) {
      out->emplace_back(ph, (int64_t)seed);
    } else {
      ok = false;
    }
    Py_DECREF(item);
  }
  Py_XDECREF(iter);
  Py_DECREF(r);
  if (PyErr_Occurred() || !ok) {
    *err_msg = py_error_string();
    return false;
  }
  return true;
}

// optional prosody DSP on f32 samples (python sonic-equivalent)
bool apply_prosody_py(const SynthesisParams &params, long sample_rate,
                      std::vector<float> *samples, std::string *err_msg) {
  if (!params.rate && !params.volume && !params.pitch) return true;
  GIL gil;
  PyObject *args = Py_BuildValue(
      "(y#iiii)", (const char *)samples->data(),
      (Py_ssize_t)(samples->size() * sizeof(float)), (int)sample_rate,
      (int)params.rate, (int)params.volume, (int)params.pitch);
  PyObject *r = args ? bridge_call("apply_prosody", args) : nullptr;
  Py_XDECREF(args);
  if (!r) {
    *err_msg = py_error_string();
    return false;
  }
  char *buf = nullptr;
  Py_ssize_t len = 0;
  if (PyBytes_AsStringAndSize(r, &buf, &len) == 0) {
    samples->assign((const float *)buf, (const float *)(buf + len));
  }
  Py_DECREF(r);
  return true;
}

// emit one SPEECH event outside the GIL; returns false on cancel
bool emit_bytes(const SynthesisParams &params, const void *data, size_t n) {
  if (!params.callback) return true;
  SynthesisEvent ev{SYNTH_EVENT_SPEECH, nullptr, (int64_t)n,
                    (uint8_t *)const_cast<void *>(data)};
  return params.callback(ev) == 0;
}

void emit_error(const SynthesisParams &params, ErrorCode code,
                const std::string &msg) {
  if (!params.callback) return;
  ExternError err;
  set_error(&err, code, msg);
  SynthesisEvent ev{SYNTH_EVENT_ERROR, &err, 0, nullptr};
  params.callback(ev);
  std::free(err.message);
}

// synthesize one sentence to f32 samples (engine lock held by caller)
std::vector<float> engine_one_shot(VoiceHandle *h, const std::string &ph,
                                   int64_t seed) {
  auto ids_v = h->engine->phonemes_to_ids(ph);
  long T = (long)ids_v.size();
  auto ids = torch::from_blob(ids_v.data(), {1, T}, torch::kLong).clone();
  auto lengths = torch::full({1}, T, torch::kLong);
  c10::optional<torch::Tensor> sid;
  if (h->num_speakers > 1)
    sid = torch::full({1}, (long)h->speaker, torch::kLong);
  auto r = h->engine->infer(ids, lengths, sid, h->noise_scale,
                            h->length_scale, h->noise_w, {seed});
  long n = r.second[0].item<long>();
  auto a = r.first.narrow(2, 0, n).to(torch::kFloat32).to(torch::kCPU)
               .contiguous();
  const float *p = a.data_ptr<float>();
  return std::vector<float>(p, p + n);
}

// native realtime: encoder once, chunked HiFi-GAN decode with the
// adaptive plan (models/chunker.py semantics; reference
// AdaptiveMelChunker piper/src/lib.rs:860-913) and 42-sample seam
// crossfade with exact timeline preservation (models/voice.py
// _stream_decode).  Calls sink(chunk) per audio chunk; returns false
// if the sink cancels.
bool engine_stream(VoiceHandle *h, const std::string &ph, int64_t seed,
                   long chunk_size, long chunk_pad,
                   const std::function<bool(std::vector<float> &)> &sink) {
  auto ids_v = h->engine->phonemes_to_ids(ph);
  long T = (long)ids_v.size();
  auto ids = torch::from_blob(ids_v.data(), {1, T}, torch::kLong).clone();
  auto lengths = torch::full({1}, T, torch::kLong);
  c10::optional<torch::Tensor> sid;
  if (h->num_speakers > 1)
    sid = torch::full({1}, (long)h->speaker, torch::kLong);
  auto enc = h->engine->infer_encoder(ids, lengths, sid, h->noise_scale,
                                      h->length_scale, h->noise_w, {seed});
  auto z = std::get<0>(enc);
  auto y_mask = std::get<1>(enc);
  auto gv = std::get<2>(enc);
  c10::optional<torch::Tensor> gopt;
  if (gv.defined() && gv.numel()) gopt = gv;
  const long F = z.size(2), hop = h->engine->config().hop();
  const long MINC = 44, MAXC = 1024, XFADE = 42;
  const bool oneshot = F <= chunk_size * 2 + chunk_pad * 2;
  long start = 0, step = 1;
  std::vector<float> tail;
  long prev_ext = 0;
  while (start < F) {
    long lo = 0, hi = F, pl = 0, pr = 0;
    bool last = true;
    if (!oneshot) {
      long size = std::min(chunk_size * step, MAXC);
      ++step;
      long end = std::min(start + size, F);
      if (F - end < MINC) end = F;
      pl = std::min(chunk_pad, start);
      pr = std::min(chunk_pad, F - end);
      last = end >= F;
      lo = start - pl;
      hi = end + pr;
      start = end;
    } else {
      start = F;
    }
    auto zc = z.narrow(2, lo, hi - lo).contiguous();
    auto mc = y_mask.narrow(2, lo, hi - lo).contiguous();
    auto a = h->engine->decode(zc, mc, gopt, c10::nullopt)
                 .to(torch::kFloat32).to(torch::kCPU).contiguous();
    const float *ptr = a[0][0].data_ptr<float>();
    const long n = a.size(2);
    const long s_lo = pl * hop;
    const long s_hi = n - pr * hop;
    const long ext = last ? 0 : std::min(XFADE, pr * hop);
    std::vector<float> cur(ptr + s_lo, ptr + s_hi + ext);
    if (!tail.empty()) {
      std::vector<float> joined = tail;
      sonata_capi::crossfade_append(joined, cur, (size_t)prev_ext);
      cur.swap(joined);
    }
    const long cut = (long)cur.size() - ext;
    std::vector<float> emitv(cur.begin(), cur.begin() + cut);
    if (!sink(emitv)) return false;
    tail.assign(cur.begin() + cut, cur.end());
    prev_ext = ext;
    if (last) break;
  }
  return true;
}

// Run the synthesis loop: iterate bridge.speak_chunks(...), firing the
// callback per chunk; FINISHED at the end; ERROR + event on failure.
void do_speak(VoiceHandle *h, std::string text, SynthesisParams params) {
  GIL gil;
  PyObject *args = Py_BuildValue(
      "(Osiiiii)", h->synth, text.c_str(), (int)params.mode,
      (int)params.rate, (int)params.volume, (int)params.pitch,
      (int)params.appended_silence_ms);
  PyObject *gen = args ? bridge_call("speak_chunks", args) : nullptr;
  Py_XDECREF(args);
  if (!gen) {
    if (params.callback) {
      ExternError err;
      set_error(&err, OPERATION_ERROR, py_error_string());
      SynthesisEvent ev{SYNTH_EVENT_ERROR, &err, 0, nullptr};
      params.callback(ev);
      std::free(err.message);
    }
    return;
  }
  PyObject *iter = PyObject_GetIter(gen);
  Py_DECREF(gen);
  bool cancelled = false;
  while (iter) {
    PyObject *item = PyIter_Next(iter);
    if (!item) break;
    char *buf = nullptr;
    Py_ssize_t len = 0;
    if (PyBytes_AsStringAndSize(item, &buf, &len) == 0 && params.callback) {
      SynthesisEvent ev{SYNTH_EVENT_SPEECH, nullptr, (int64_t)len,
                        reinterpret_cast<uint8_t *>(buf)};
      uint8_t rc;
      {
        // release the GIL while user code runs
        PyThreadState *ts = PyEval_SaveThread();
        rc = params.callback(ev);
        PyEval_RestoreThread(ts);
      }
      if (rc != 0) cancelled = true;
    }
    Py_DECREF(item);
    if (cancelled) break;
  }
  Py_XDECREF(iter);
  if (PyErr_Occurred()) {
    std::string msg = py_error_string();
    if (params.callback) {
      ExternError err;
      set_error(&err, classify(msg), msg);
      SynthesisEvent ev{SYNTH_EVENT_ERROR, &err, 0, nullptr};
      params.callback(ev);
      std::free(err.message);
    }
    return;
  }
  if (params.callback && !cancelled) {
    SynthesisEvent ev{SYNTH_EVENT_FINISHED, nullptr, 0, nullptr};
    params.callback(ev);
  }
}

// native synthesis loop: GIL only for phonemize + optional prosody
void do_speak_native(VoiceHandle *h, std::string text,
                     SynthesisParams params) {
  std::vector<std::pair<std::string, int64_t>> sents;
  std::string msg;
  if (!frontend_phonemize(h, text, h->speaker, &sents, &msg)) {
    emit_error(params, classify(msg), msg);
    return;
  }
  torch::NoGradGuard ng;
  std::unique_lock<std::mutex> lk(h->mu);
  bool cancelled = false;
  try {
    for (auto &ps : sents) {
      const std::string &ph = ps.first;
      const int64_t seed = ps.second;
      if (params.mode == SYNTH_MODE_REALTIME) {
        bool ok = engine_stream(
            h, ph, seed, 72, 3,  // reference capi chunking (:407-409)
            [&](std::vector<float> &chunk) {
              std::string perr;
              if (!apply_prosody_py(params, h->sample_rate, &chunk, &perr))
                return false;
              auto pcm = sonata_capi::to_i16(chunk.data(), chunk.size());
              return emit_bytes(params, pcm.data(), pcm.size() * 2);
            });
        if (!ok) {
          cancelled = true;
          break;
        }
      } else {  // LAZY / PARALLEL: per-sentence one-shot
        auto samples = engine_one_shot(h, ph, seed);
        std::string perr;
        if (!apply_prosody_py(params, h->sample_rate, &samples, &perr)) {
          emit_error(params, OPERATION_ERROR, perr);
          return;
        }
        auto pcm = sonata_capi::to_i16(samples.data(), samples.size());
        if (!emit_bytes(params, pcm.data(), pcm.size() * 2)) {
          cancelled = true;
          break;
        }
      }
      if (params.appended_silence_ms && !cancelled) {
        std::vector<int16_t> sil(
            (size_t)(h->sample_rate * params.appended_silence_ms / 1000), 0);
        if (!emit_bytes(params, sil.data(), sil.size() * 2)) {
          cancelled = true;
          break;
        }
      }
    }
  } catch (const std::exception &e) {
    emit_error(params, OPERATION_ERROR, e.what());
    return;
  }
  if (params.callback && !cancelled) {
    SynthesisEvent ev{SYNTH_EVENT_FINISHED, nullptr, 0, nullptr};
    params.callback(ev);
  }
}

}  // namespace

extern "C" {

void libsonataFreeString(int8_t *string_ptr) {
  std::free(string_ptr);
}

void libsonataFreePiperSynthConfig(PiperSynthConfig *synth_config) {
  std::free(synth_config);
}

void libsonataFreeSynthesisEvent(SynthesisEvent event) {
  (void)event;  // event data is owned by the engine; nothing to free
}

SonataVoice *libsonataLoadVoiceFromConfigPath(FfiStr config_path_ptr,
                                              ExternError *out_error) {
  ensure_interpreter();
  set_success(out_error);
  if (!config_path_ptr) {
    set_error(out_error, INVALID_UTF8_SEQUENCE, "null config path");
    return nullptr;
  }
  auto *h = new VoiceHandle();
  // 1) native engine (hot path): C++ loader, no Python involved
  try {
    const char *dev_env = std::getenv("SONATA_DEVICE");
    std::string dev = dev_env && *dev_env
                          ? dev_env
                          : (torch::cuda::is_available() ? "cuda:0" : "cpu");
    torch::Device device(dev);
    torch::Dtype dtype =
        device.is_cuda() ? torch::kBFloat16 : torch::kFloat32;
    h->engine = new sonata::VitsEngine(config_path_ptr, device, dtype);
    const auto &cfg = h->engine->config();
    h->sample_rate = cfg.sample_rate;
    h->num_speakers = cfg.num_speakers;
    h->length_scale = (float)cfg.length_scale;
    h->noise_scale = (float)cfg.noise_scale;
    h->noise_w = (float)cfg.noise_w;
  } catch (const std::exception &e) {
    h->engine = nullptr;
  }
  GIL gil;
  if (h->engine) {
    // text front-end only (no weights)
    PyObject *args = Py_BuildValue("(s)", config_path_ptr);
    h->frontend = args ? bridge_call("load_frontend", args) : nullptr;
    Py_XDECREF(args);
    if (!h->frontend) {
      // cannot phonemize -> fall back entirely to the python path
      delete h->engine;
      h->engine = nullptr;
      PyErr_Clear();
    }
  }
  if (!h->engine) {
    PyObject *args = Py_BuildValue("(s)", config_path_ptr);
    h->synth = args ? bridge_call("load_voice", args) : nullptr;
    Py_XDECREF(args);
    if (!h->synth) {
      std::string msg = py_error_string();
      set_error(out_error, FAILED_TO_LOAD_RESOURCE, msg);
      delete h;
      return nullptr;
    }
  }
  return reinterpret_cast<SonataVoice *>(h);
}

void libsonataUnloadSonataVoice(SonataVoice *voice_ptr) {
  if (!voice_ptr) return;
  auto *h = reinterpret_cast<VoiceHandle *>(voice_ptr);
  delete h->engine;
  {
    GIL gil;
    Py_XDECREF(h->synth);
    Py_XDECREF(h->frontend);
  }
  delete h;
}

void libsonataGetAudioInfo(SonataVoice *voice_ptr, AudioInfo *audio_info_ptr,
                           ExternError *out_error) {
  set_success(out_error);
  if (!voice_ptr || !audio_info_ptr) {
    set_error(out_error, ErrorCode_INVALID_HANDLE, "null handle");
    return;
  }
  auto *h = reinterpret_cast<VoiceHandle *>(voice_ptr);
  if (h->engine) {  // native: no GIL
    audio_info_ptr->sample_rate = (uint32_t)h->sample_rate;
    audio_info_ptr->num_channels = 1;
    audio_info_ptr->sample_width = 2;
    return;
  }
  GIL gil;
  PyObject *args = Py_BuildValue("(O)", h->synth);
  PyObject *r = args ? bridge_call("get_audio_info", args) : nullptr;
  Py_XDECREF(args);
  if (!r) {
    set_error(out_error, OPERATION_ERROR, py_error_string());
    return;
  }
  unsigned int sr = 0, ch = 0, w = 0;
  if (PyArg_ParseTuple(r, "III", &sr, &ch, &w)) {
    audio_info_ptr->sample_rate = sr;
    audio_info_ptr->num_channels = ch;
    audio_info_ptr->sample_width = w;
  } else {
    set_error(out_error, OPERATION_ERROR, py_error_string());
  }
  Py_DECREF(r);
}

PiperSynthConfig *libsonataGetPiperDefaultSynthConfig(
    SonataVoice *voice_ptr, ExternError *out_error) {
  set_success(out_error);
  if (!voice_ptr) {
    set_error(out_error, ErrorCode_INVALID_HANDLE, "null handle");
    return nullptr;
  }
  auto *h = reinterpret_cast<VoiceHandle *>(voice_ptr);
  if (h->engine) {  // native: no GIL
    PiperSynthConfig *cfg =
        static_cast<PiperSynthConfig *>(std::malloc(sizeof(PiperSynthConfig)));
    std::lock_guard<std::mutex> lk(h->mu);
    cfg->speaker = h->speaker;
    cfg->length_scale = h->length_scale;
    cfg->noise_scale = h->noise_scale;
    cfg->noise_w = h->noise_w;
    return cfg;
  }
  GIL gil;
  PyObject *args = Py_BuildValue("(O)", h->synth);
  PyObject *r = args ? bridge_call("get_synth_config", args) : nullptr;
  Py_XDECREF(args);
  if (!r) {
    set_error(out_error, OPERATION_ERROR, py_error_string());
    return nullptr;
  }
  unsigned int speaker = 0;
  float ls = 0, ns = 0, nw = 0;
  PiperSynthConfig *cfg = nullptr;
  if (PyArg_ParseTuple(r, "Ifff", &speaker, &ls, &ns, &nw)) {
    cfg = static_cast<PiperSynthConfig *>(std::malloc(sizeof(*cfg)));
    cfg->speaker = speaker;
    cfg->length_scale = ls;
    cfg->noise_scale = ns;
    cfg->noise_w = nw;
  } else {
    set_error(out_error, OPERATION_ERROR, py_error_string());
  }
  Py_DECREF(r);
  return cfg;
}

void libsonataSetPiperSynthConfig(SonataVoice *voice_ptr,
                                  PiperSynthConfig synth_config,
                                  ExternError *out_error) {
  set_success(out_error);
  if (!voice_ptr) {
    set_error(out_error, ErrorCode_INVALID_HANDLE, "null handle");
    return;
  }
  auto *h = reinterpret_cast<VoiceHandle *>(voice_ptr);
  if (h->engine) {  // native: no GIL
    std::lock_guard<std::mutex> lk(h->mu);
    h->speaker = synth_config.speaker;
    h->length_scale = synth_config.length_scale;
    h->noise_scale = synth_config.noise_scale;
    h->noise_w = synth_config.noise_w;
    return;
  }
  GIL gil;
  PyObject *args = Py_BuildValue(
      "(OIfff)", h->synth, (unsigned int)synth_config.speaker,
      (double)synth_config.length_scale, (double)synth_config.noise_scale,
      (double)synth_config.noise_w);
  PyObject *r = args ? bridge_call("set_synth_config", args) : nullptr;
  Py_XDECREF(args);
  if (!r) {
    set_error(out_error, OPERATION_ERROR, py_error_string());
    return;
  }
  Py_DECREF(r);
}

void libsonataSpeak(SonataVoice *voice_ptr, FfiStr text_ptr,
                    SynthesisParams params, ExternError *out_error) {
  set_success(out_error);
  if (!voice_ptr || !text_ptr) {
    set_error(out_error, ErrorCode_INVALID_HANDLE, "null handle/text");
    return;
  }
  if (params.mode < SYNTH_MODE_LAZY || params.mode > SYNTH_MODE_REALTIME) {
    set_error(out_error, INVALID_SYNTHESIS_MODE, "invalid synthesis mode");
    return;
  }
  auto *h = reinterpret_cast<VoiceHandle *>(voice_ptr);
  std::string text(text_ptr);
  auto run = [h, text, params] {
    if (h->engine) {
      do_speak_native(h, text, params);
    } else {
      do_speak(h, text, params);
    }
  };
  if (params.nonblocking) {
    std::thread(run).detach();
  } else {
    run();
  }
}

uint8_t libsonataIsNativeEngine(SonataVoice *voice_ptr) {
  if (!voice_ptr) return 0;
  return reinterpret_cast<VoiceHandle *>(voice_ptr)->engine ? 1 : 0;
}

uint8_t libsonataSpeakToFile(SonataVoice *voice_ptr, FfiStr text_ptr,
                             SynthesisParams params, FfiStr out_filename_ptr,
                             ExternError *out_error) {
  set_success(out_error);
  if (!voice_ptr || !text_ptr || !out_filename_ptr) {
    set_error(out_error, ErrorCode_INVALID_HANDLE, "null handle/text/path");
    return 0;
  }
  auto *h = reinterpret_cast<VoiceHandle *>(voice_ptr);
  if (h->engine) {  // native: synthesize all sentences, write WAV
    std::vector<std::pair<std::string, int64_t>> sents;
    std::string msg;
    if (!frontend_phonemize(h, text_ptr, h->speaker, &sents, &msg)) {
      set_error(out_error, classify(msg), msg);
      return 0;
    }
    try {
      torch::NoGradGuard ng;
      std::unique_lock<std::mutex> lk(h->mu);
      std::vector<float> all;
      for (auto &ps : sents) {
        auto s = engine_one_shot(h, ps.first, ps.second);
        std::string perr;
        if (!apply_prosody_py(params, h->sample_rate, &s, &perr)) {
          set_error(out_error, OPERATION_ERROR, perr);
          return 0;
        }
        all.insert(all.end(), s.begin(), s.end());
        if (params.appended_silence_ms)
          all.insert(all.end(),
                     (size_t)(h->sample_rate * params.appended_silence_ms /
                              1000),
                     0.0f);
      }
      auto pcm = sonata_capi::to_i16(all.data(), all.size());
      auto wav = sonata_capi::wav_bytes(pcm, (uint32_t)h->sample_rate);
      std::ofstream f(out_filename_ptr, std::ios::binary);
      if (!f.good()) {
        set_error(out_error, OPERATION_ERROR,
                  std::string("cannot open ") + out_filename_ptr);
        return 0;
      }
      f.write((const char *)wav.data(), (std::streamsize)wav.size());
      return 1;
    } catch (const std::exception &e) {
      set_error(out_error, OPERATION_ERROR, e.what());
      return 0;
    }
  }
  GIL gil;
  PyObject *args = Py_BuildValue(
      "(Ossiiii)", h->synth, text_ptr, out_filename_ptr, (int)params.rate,
      (int)params.volume, (int)params.pitch,
      (int)params.appended_silence_ms);
  PyObject *r = args ? bridge_call("speak_to_file", args) : nullptr;
  Py_XDECREF(args);
  if (!r) {
    std::string msg = py_error_string();
    set_error(out_error, classify(msg), msg);
    return 0;
  }
  Py_DECREF(r);
  return 1;
}

}  // extern "C"
