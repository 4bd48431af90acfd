// libsonata_amd C ABI implementation: embeds CPython and drives the
// sonata_amd engine through sonata_amd.frontends.capi_bridge.
//
// Behavior parity with the reference C API (crates/frontends/capi/src/
// lib.rs): opaque voice handles (:40-64), error codes (:19-26), chunked
// callback protocol with SPEECH/FINISHED/ERROR events and nonzero-return
// cancellation (:415-438), nonblocking dispatch on a worker thread
// (:366-386), realtime chunking handled bridge-side (72/3, :407-409).
//
// Threading: every entry point (and the nonblocking worker) brackets its
// Python calls with PyGILState_Ensure/Release, so the library works both
// from plain C programs (it initializes the interpreter on first use)
// and inside an existing Python process (ctypes).
#include "libsonata_amd.h"

#include <Python.h>

#include <cstdlib>
#include <cstring>
#include <mutex>
#include <string>
#include <thread>

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
  PyObject *synth;  // SonataSpeechSynthesizer
};

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
  GIL gil;
  set_success(out_error);
  if (!config_path_ptr) {
    set_error(out_error, INVALID_UTF8_SEQUENCE, "null config path");
    return nullptr;
  }
  PyObject *args = Py_BuildValue("(s)", config_path_ptr);
  PyObject *synth = args ? bridge_call("load_voice", args) : nullptr;
  Py_XDECREF(args);
  if (!synth) {
    std::string msg = py_error_string();
    set_error(out_error, FAILED_TO_LOAD_RESOURCE, msg);
    return nullptr;
  }
  auto *h = new VoiceHandle{synth};
  return reinterpret_cast<SonataVoice *>(h);
}

void libsonataUnloadSonataVoice(SonataVoice *voice_ptr) {
  if (!voice_ptr) return;
  auto *h = reinterpret_cast<VoiceHandle *>(voice_ptr);
  {
    GIL gil;
    Py_XDECREF(h->synth);
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
  if (params.nonblocking) {
    std::thread([h, text, params] { do_speak(h, text, params); }).detach();
  } else {
    do_speak(h, text, params);
  }
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
