/* libsonata_amd — C ABI for the MI355X-native Sonata engine.
 *
 * Mirrors the reference C API surface (crates/frontends/capi/libsonata.h:
 * error codes 16-21, event types SPEECH/FINISHED/ERROR, modes
 * LAZY/PARALLEL/REALTIME, PiperSynthConfig / AudioInfo / SynthesisParams /
 * SynthesisEvent structs, libsonata* entry points) so C callers of the
 * reference can switch by relinking.  The implementation embeds CPython
 * and drives the sonata_amd engine (GPU path when a MI355X is visible).
 */
#ifndef LIBSONATA_AMD_H
#define LIBSONATA_AMD_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

#define INVALID_SYNTHESIS_MODE 16
#define FAILED_TO_LOAD_RESOURCE 17
#define PHONEMIZATION_ERROR 18
#define OPERATION_ERROR 19
#define INVALID_UTF8_SEQUENCE 20
#define UNKNOWN_ERROR 21

#define SYNTH_EVENT_SPEECH 0
#define SYNTH_EVENT_FINISHED 1
#define SYNTH_EVENT_ERROR 2

#define SYNTH_MODE_LAZY 0
#define SYNTH_MODE_PARALLEL 1
#define SYNTH_MODE_REALTIME 2

typedef struct SonataVoice SonataVoice;

typedef struct PiperSynthConfig {
  uint32_t speaker;
  float length_scale;
  float noise_scale;
  float noise_w;
} PiperSynthConfig;

typedef int32_t ErrorCode;
#define ErrorCode_SUCCESS 0
#define ErrorCode_PANIC -1
#define ErrorCode_INVALID_HANDLE -1000

typedef struct ExternError {
  ErrorCode code;
  char *message; /* owned; release with libsonataFreeString */
} ExternError;

typedef struct SynthesisEvent {
  int32_t event_type;
  struct ExternError *error_ptr;
  int64_t len;
  uint8_t *data;
} SynthesisEvent;

typedef const char *FfiStr;

typedef struct AudioInfo {
  uint32_t sample_rate;
  uint32_t num_channels;
  uint32_t sample_width;
} AudioInfo;

/* Return nonzero to cancel synthesis. */
typedef uint8_t (*SpeechSynthesisCallback)(struct SynthesisEvent);

typedef struct SynthesisParams {
  int32_t mode;
  uint8_t rate;    /* 0 = unset, else percent 1-100 */
  uint8_t volume;
  uint8_t pitch;
  uint32_t appended_silence_ms;
  SpeechSynthesisCallback callback;
  uint8_t nonblocking;
} SynthesisParams;

void libsonataFreeString(int8_t *string_ptr);
void libsonataFreePiperSynthConfig(struct PiperSynthConfig *synth_config);
void libsonataFreeSynthesisEvent(struct SynthesisEvent event);

struct SonataVoice *libsonataLoadVoiceFromConfigPath(
    FfiStr config_path_ptr, struct ExternError *out_error);
void libsonataUnloadSonataVoice(struct SonataVoice *voice_ptr);
void libsonataGetAudioInfo(struct SonataVoice *voice_ptr,
                           struct AudioInfo *audio_info_ptr,
                           struct ExternError *out_error);
struct PiperSynthConfig *libsonataGetPiperDefaultSynthConfig(
    struct SonataVoice *voice_ptr, struct ExternError *out_error);
void libsonataSetPiperSynthConfig(struct SonataVoice *voice_ptr,
                                  struct PiperSynthConfig synth_config,
                                  struct ExternError *out_error);
void libsonataSpeak(struct SonataVoice *voice_ptr, FfiStr text_ptr,
                    struct SynthesisParams params,
                    struct ExternError *out_error);
/* 1 when the voice runs on the native C++ engine (GIL-free synthesis),
 * 0 when it fell back to the Python bridge.  sonata_amd extension (not
 * in the reference API). */
uint8_t libsonataIsNativeEngine(struct SonataVoice *voice_ptr);

uint8_t libsonataSpeakToFile(struct SonataVoice *voice_ptr, FfiStr text_ptr,
                             struct SynthesisParams params,
                             FfiStr out_filename_ptr,
                             struct ExternError *out_error);

#ifdef __cplusplus
}
#endif
#endif /* LIBSONATA_AMD_H */
