// sonata_infer — standalone C++ CLI over the VitsEngine runtime.
//
// Reads IPA phoneme lines (one sentence per line) from a file or stdin,
// synthesizes each through the C++ engine (GPU bf16 when available) and
// writes 16-bit PCM WAV files.  The native-binary counterpart of the
// Python CLI (sonata_amd/frontends/cli.py); no Python interpreter
// involved anywhere in this path.
//
// Usage:
//   sonata_infer <voice.json> [-o out.wav] [-d cuda:0|cpu] [-f phonemes.txt]
//                [--length-scale F] [--noise-scale F] [--noise-w F]
//                [--speaker N] [--bench N] [--stream [chunk] [pad]]
#include <chrono>
#include <cstdlib>
#include <cstring>
#include <fstream>
#include <iostream>
#include <sstream>
#include <string>
#include <vector>

#include "vits_engine.h"

namespace {

void write_wav(const std::string& path, const float* samples, size_t n,
               long sample_rate) {
  // peak-normalizing f32 -> i16 (audio-ops samples.rs:51-75 semantics)
  float peak = 1e-6f;
  for (size_t i = 0; i < n; ++i) peak = std::max(peak, std::abs(samples[i]));
  const float scale = peak > 1.f ? 32767.f / peak : 32767.f;
  std::vector<int16_t> pcm(n);
  for (size_t i = 0; i < n; ++i) {
    float v = samples[i] * scale;
    pcm[i] = (int16_t)std::max(-32768.f, std::min(32767.f, v));
  }
  std::ofstream f(path, std::ios::binary);
  uint32_t data_bytes = (uint32_t)(n * 2);
  uint32_t chunk = 36 + data_bytes;
  uint32_t byte_rate = (uint32_t)(sample_rate * 2);
  uint16_t block_align = 2, bits = 16, fmt = 1, channels = 1;
  uint32_t sr = (uint32_t)sample_rate, fmt_size = 16;
  f.write("RIFF", 4);
  f.write((char*)&chunk, 4);
  f.write("WAVE", 4);
  f.write("fmt ", 4);
  f.write((char*)&fmt_size, 4);
  f.write((char*)&fmt, 2);
  f.write((char*)&channels, 2);
  f.write((char*)&sr, 4);
  f.write((char*)&byte_rate, 4);
  f.write((char*)&block_align, 2);
  f.write((char*)&bits, 2);
  f.write("data", 4);
  f.write((char*)&data_bytes, 4);
  f.write((char*)pcm.data(), data_bytes);
}

uint64_t fnv1a(const std::string& s) {
  uint64_t h = 1469598103934665603ull;
  for (unsigned char c : s) { h ^= c; h *= 1099511628211ull; }
  return h & 0x7fffffffffffffffull;
}

// The HIP runtime\'s static destructors segfault at process exit when
// torch_hip is loaded (observed on ROCm 7.2: __hip_module_dtor inside
// __cxa_finalize).  All useful work is done by then; exit without
// running static dtors.
[[noreturn]] void clean_exit(int code) {
  std::cout.flush();
  std::cerr.flush();
  std::_Exit(code);
}

}  // namespace

int main(int argc, char** argv) {
  if (argc < 2) {
    std::cerr << "usage: sonata_infer <voice.json> [-o out.wav] [-d device]"
                 " [-f phonemes.txt] [--bench N]\n";
    return 2;
  }
  std::string config = argv[1], out = "out.wav", device_s, input_file;
  double ls = 0, ns = 0, nw = 0;
  long speaker = -1, bench = 0, stream_chunk = 0, stream_pad = 3;
  for (int i = 2; i < argc; ++i) {
    std::string a = argv[i];
    auto next = [&]() { return std::string(argv[++i]); };
    if (a == "-o") out = next();
    else if (a == "-d") device_s = next();
    else if (a == "-f") input_file = next();
    else if (a == "--length-scale") ls = std::stod(next());
    else if (a == "--noise-scale") ns = std::stod(next());
    else if (a == "--noise-w") nw = std::stod(next());
    else if (a == "--speaker") speaker = std::stol(next());
    else if (a == "--bench") bench = std::stol(next());
    else if (a == "--stream") stream_chunk = 45;
    else if (a == "--stream-chunk") stream_chunk = std::stol(next());
    else if (a == "--stream-pad") stream_pad = std::stol(next());
  }
  if (device_s.empty())
    device_s = torch::cuda::is_available() ? "cuda:0" : "cpu";
  torch::Device device(device_s);
  torch::Dtype dtype =
      device.is_cuda() ? torch::kBFloat16 : torch::kFloat32;

  sonata::VitsEngine engine(config, device, dtype);
  const auto& cfg = engine.config();
  if (ls == 0) ls = cfg.length_scale;
  if (ns == 0) ns = cfg.noise_scale;
  if (nw == 0) nw = cfg.noise_w;
  std::cerr << "sonata_infer: voice loaded on " << device_s << " (sr "
            << cfg.sample_rate << ", " << cfg.num_speakers
            << " speaker(s))\n";

  std::vector<std::string> sentences;
  std::istream* in = &std::cin;
  std::ifstream fin;
  if (!input_file.empty()) {
    fin.open(input_file);
    if (!fin.good()) {
      std::cerr << "cannot open " << input_file << "\n";
      return 1;
    }
    in = &fin;
  }
  std::string line;
  while (std::getline(*in, line))
    if (!line.empty()) sentences.push_back(line);
  if (sentences.empty()) {
    std::cerr << "no input phoneme lines\n";
    return 1;
  }

  // encode batch
  std::vector<std::vector<int64_t>> id_lists;
  long T = 1;
  for (auto& s : sentences) {
    id_lists.push_back(engine.phonemes_to_ids(s));
    T = std::max(T, (long)id_lists.back().size());
  }
  long B = (long)id_lists.size();
  auto ids = torch::zeros({B, T}, torch::kLong);
  auto lengths = torch::zeros({B}, torch::kLong);
  std::vector<int64_t> seeds;
  for (long b = 0; b < B; ++b) {
    auto& il = id_lists[b];
    for (size_t t = 0; t < il.size(); ++t) ids[b][t] = il[t];
    lengths[b] = (long)il.size();
    seeds.push_back((int64_t)fnv1a(sentences[b]));
  }
  c10::optional<torch::Tensor> sid;
  if (speaker >= 0)
    sid = torch::full({B}, speaker, torch::kLong);

  if (stream_chunk > 0) {
    // Native streaming: encoder once, HiFi-GAN decoded in adaptive
    // chunks (ports models/chunker.py: growth x step, MIN 44 / MAX 1024
    // frames, overlap-discard +-pad frames; reference AdaptiveMelChunker
    // semantics, piper/src/lib.rs:860-913).  One utterance at a time.
    std::vector<float> samples;
    double first_ms = -1;
    auto s0 = std::chrono::steady_clock::now();
    for (long b = 0; b < B; ++b) {
      auto idsb = ids.narrow(0, b, 1);
      auto lensb = lengths.narrow(0, b, 1);
      std::vector<int64_t> sb{seeds[(size_t)b]};
      auto [z, y_mask, gv] = engine.infer_encoder(idsb, lensb, sid, ns, ls,
                                                  nw, sb);
      c10::optional<torch::Tensor> gopt;
      if (gv.defined() && gv.numel()) gopt = gv;
      const long F = z.size(2), hop = engine.config().hop();
      const long MINC = 44, MAXC = 1024;
      long start = 0, step = 1;
      bool oneshot = F <= stream_chunk * 2 + stream_pad * 2;
      while (start < F) {
        long lo = 0, hi = F, pl = 0, pr = 0;
        bool last = true;
        if (!oneshot) {
          long size = std::min(stream_chunk * step, MAXC);
          ++step;
          long end = std::min(start + size, F);
          if (F - end < MINC) end = F;
          pl = std::min(stream_pad, start);
          pr = std::min(stream_pad, F - end);
          last = end >= F;
          lo = start - pl;
          hi = end + pr;
          start = end;
        } else {
          start = F;
        }
        auto zc = z.narrow(2, lo, hi - lo).contiguous();
        auto mc = y_mask.narrow(2, lo, hi - lo).contiguous();
        auto a = engine.decode(zc, mc, gopt, c10::nullopt)
                     .to(torch::kFloat32).to(torch::kCPU).contiguous();
        const float* ptr = a[0][0].data_ptr<float>();
        long n = a.size(2);
        samples.insert(samples.end(), ptr + pl * hop,
                       ptr + (n - pr * hop));
        if (first_ms < 0)
          first_ms = std::chrono::duration<double, std::milli>(
                         std::chrono::steady_clock::now() - s0).count();
        if (last) break;
      }
    }
    write_wav(out, samples.data(), samples.size(), cfg.sample_rate);
    double dur_ms = 1000.0 * samples.size() / cfg.sample_rate;
    std::cerr << "streamed " << out << ": " << samples.size()
              << " samples (" << dur_ms << " ms audio), first chunk in "
              << first_ms << " ms\n";
    clean_exit(0);
  }

  auto t0 = std::chrono::steady_clock::now();
  auto [audio, audio_lengths] = engine.infer(ids, lengths, sid, ns, ls, nw,
                                             seeds);
  if (device.is_cuda()) torch::cuda::synchronize();
  auto t1 = std::chrono::steady_clock::now();
  double infer_ms =
      std::chrono::duration<double, std::milli>(t1 - t0).count();

  if (bench > 0) {
    for (long i = 0; i < 2; ++i)
      engine.infer(ids, lengths, sid, ns, ls, nw, seeds);
    if (device.is_cuda()) torch::cuda::synchronize();
    auto b0 = std::chrono::steady_clock::now();
    double audio_sec = 0;
    for (long i = 0; i < bench; ++i) {
      auto r = engine.infer(ids, lengths, sid, ns, ls, nw, seeds);
      audio_sec +=
          r.second.sum().item<double>() / (double)cfg.sample_rate;
    }
    if (device.is_cuda()) torch::cuda::synchronize();
    auto b1 = std::chrono::steady_clock::now();
    double el = std::chrono::duration<double>(b1 - b0).count();
    std::cout << "{\"bench_steps\": " << bench << ", \"audio_sec_per_s\": "
              << audio_sec / el << ", \"ms_per_step\": "
              << el * 1000.0 / bench << "}\n";
    clean_exit(0);
  }

  auto af = audio.to(torch::kFloat32).to(torch::kCPU).contiguous();
  auto al = audio_lengths.to(torch::kCPU);
  // concatenate valid samples of all sentences into one file
  std::vector<float> samples;
  for (long b = 0; b < B; ++b) {
    long n = al[b].item<long>();
    const float* ptr = af[b][0].data_ptr<float>();
    samples.insert(samples.end(), ptr, ptr + n);
  }
  write_wav(out, samples.data(), samples.size(), cfg.sample_rate);
  double dur_ms = 1000.0 * samples.size() / cfg.sample_rate;
  std::cerr << "wrote " << out << ": " << samples.size() << " samples ("
            << dur_ms << " ms audio), infer " << infer_ms << " ms, rtf "
            << infer_ms / dur_ms << "\n";
  clean_exit(0);
}
