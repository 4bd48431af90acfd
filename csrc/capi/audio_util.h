// Native audio utilities for the C ABI hot path: f32 -> i16 peak
// normalization and the stream-seam crossfade, mirroring
// sonata_amd/audio/samples.py (reference semantics: audio-ops
// samples.rs:51-75 to_i16 scaling, :144-157 quarter-sine crossfade).
#pragma once

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <cstring>
#include <vector>

namespace sonata_capi {

// float32 -> little-endian i16 PCM with peak normalization
// (scale = 32767/absmax — amplifies quiet audio too, matching the
// reference exactly).
inline std::vector<int16_t> to_i16(const float* s, size_t n) {
  std::vector<int16_t> out(n);
  float peak = 0.f;
  for (size_t i = 0; i < n; ++i) peak = std::max(peak, std::abs(s[i]));
  const float scale = peak > 1e-8f ? 32767.0f / peak : 0.0f;
  for (size_t i = 0; i < n; ++i) {
    float v = s[i] * scale;
    out[i] = (int16_t)std::max(-32768.f, std::min(32767.f, v));
  }
  return out;
}

// quarter-sine 0->1 ramp of length n (samples.py _quarter_sine_ramp)
inline std::vector<float> qsine_ramp(size_t n) {
  std::vector<float> r(n);
  if (n == 1) {
    r[0] = 0.f;  // np.linspace(0, pi/2, 1) == [0]
    return r;
  }
  for (size_t i = 0; i < n; ++i)
    r[i] = std::sin((float)i / (float)(n - 1) * 1.5707963f);
  return r;
}

// join a and b with an n-sample equal-power crossfade (in place into a)
inline void crossfade_append(std::vector<float>& a,
                             const std::vector<float>& b, size_t n) {
  n = std::min({n, a.size(), b.size()});
  if (n == 0) {
    a.insert(a.end(), b.begin(), b.end());
    return;
  }
  auto ramp = qsine_ramp(n);
  const size_t off = a.size() - n;
  for (size_t i = 0; i < n; ++i)
    a[off + i] = a[off + i] * ramp[n - 1 - i] + b[i] * ramp[i];
  a.insert(a.end(), b.begin() + n, b.end());
}

// 16-bit PCM mono WAV (wav.py / sonata_main.cpp write_wav semantics)
inline std::vector<uint8_t> wav_bytes(const std::vector<int16_t>& pcm,
                                      uint32_t sr) {
  const uint32_t data_bytes = (uint32_t)(pcm.size() * 2);
  std::vector<uint8_t> out(44 + data_bytes);
  uint8_t* p = out.data();
  auto put32 = [&](uint32_t v) { std::memcpy(p, &v, 4); p += 4; };
  auto put16 = [&](uint16_t v) { std::memcpy(p, &v, 2); p += 2; };
  std::memcpy(p, "RIFF", 4); p += 4;
  put32(36 + data_bytes);
  std::memcpy(p, "WAVE", 4); p += 4;
  std::memcpy(p, "fmt ", 4); p += 4;
  put32(16); put16(1); put16(1); put32(sr); put32(sr * 2);
  put16(2); put16(16);
  std::memcpy(p, "data", 4); p += 4;
  put32(data_bytes);
  std::memcpy(p, pcm.data(), data_bytes);
  return out;
}

}  // namespace sonata_capi
