// VitsEngine — the MI355X-native C++ inference runtime.
//
// This is the component that replaces the reference's dependency on ONNX
// Runtime (ort executes the whole VITS graph,
// crates/sonata/models/piper/src/lib.rs:79-86): a from-scratch C++
// executor that loads a sonata_amd voice pack (<stem>.json config +
// <stem>.safetensors weights) and runs the complete Piper/VITS graph —
// text encoder (relative-position attention), stochastic duration
// predictor (rational-quadratic spline flows), residual-coupling flow,
// HiFi-GAN generator — on ATen tensors, dispatching every hot op to the
// hand-written CDNA4 HIP kernels (csrc/conv1d_cl.hip, conv1d.hip,
// elementwise.hip) when the device is a GPU, and to plain fp32 ATen ops
// on CPU (the numerics oracle).
//
// Exposed to Python (for parity tests / optional use) through ext.cpp
// and to C/C++ through the `sonata_infer` CLI (sonata_main.cpp).
#pragma once

#include <torch/torch.h>

#include <map>
#include <string>
#include <unordered_map>
#include <utility>
#include <vector>

namespace sonata {

struct EngineConfig {
  long sample_rate = 22050;
  long num_speakers = 1;
  long num_symbols = 0;
  double noise_scale = 0.667, length_scale = 1.0, noise_w = 0.8;
  // architecture
  long inter = 192, hidden = 192, filter = 768, n_heads = 2, n_layers = 6,
       kernel_size = 3, window_size = 4, gin = 0;
  std::vector<long> resblock_ks{3, 7, 11};
  std::vector<std::vector<long>> resblock_dil{{1, 3, 5}, {1, 3, 5}, {1, 3, 5}};
  std::vector<long> up_rates{8, 8, 2, 2};
  std::vector<long> up_ks{16, 16, 4, 4};
  long up_init_ch = 512;
  long hop() const {
    long h = 1;
    for (long r : up_rates) h *= r;
    return h;
  }
  // utf8 phoneme char -> ids (piper phoneme_id_map)
  std::map<std::string, std::vector<long>> phoneme_id_map;
};

class VitsEngine {
 public:
  VitsEngine(const std::string& config_path, torch::Device device,
             torch::Dtype dtype);

  const EngineConfig& config() const { return cfg_; }

  // phoneme ids: BOS/EOS wrap + PAD interleave (piper lib.rs:232-250)
  std::vector<int64_t> phonemes_to_ids(const std::string& utf8) const;

  // ids [B, T] int64, lengths [B] int64 -> (audio [B, 1, S], audio_lengths
  // [B]).  seeds: one per utterance (deterministic noise; empty = seeded
  // from std random).
  std::pair<torch::Tensor, torch::Tensor> infer(
      torch::Tensor ids, torch::Tensor lengths,
      c10::optional<torch::Tensor> sid, double noise_scale,
      double length_scale, double noise_w,
      const std::vector<int64_t>& seeds);

  // streaming split (reference encoder.onnx / decoder.onnx pair)
  std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> infer_encoder(
      torch::Tensor ids, torch::Tensor lengths,
      c10::optional<torch::Tensor> sid, double noise_scale,
      double length_scale, double noise_w,
      const std::vector<int64_t>& seeds);
  torch::Tensor decode(torch::Tensor z, torch::Tensor y_mask,
                       c10::optional<torch::Tensor> g,
                       c10::optional<torch::Tensor> lengths);

 private:
  EngineConfig cfg_;
  torch::Device device_;
  torch::Dtype dtype_;
  std::unordered_map<std::string, torch::Tensor> P_;
  mutable std::unordered_map<std::string, torch::Tensor> cache_;

  bool gpu() const { return device_.is_cuda(); }
  torch::Tensor p(const std::string& name) const;
  bool has(const std::string& name) const { return P_.count(name) > 0; }
  c10::optional<torch::Tensor> maybe(const std::string& name) const;
  torch::Tensor perm_conv(const std::string& wname) const;
  torch::Tensor perm_convt(const std::string& wname, long stride) const;
  torch::Tensor bias_f32(const std::string& bname) const;

  // op helpers (mirror sonata_amd/ops/functional.py dispatch)
  torch::Tensor conv(torch::Tensor x, const std::string& mod, long stride = 1,
                     long pad = 0, long dil = 1, long groups = 1,
                     double pre_lrelu = 0.0, double post_lrelu = 0.0) const;
  torch::Tensor layer_norm(torch::Tensor x, const std::string& mod,
                           c10::optional<torch::Tensor> residual = {}) const;
  torch::Tensor gate(torch::Tensor x, c10::optional<torch::Tensor> g,
                     long n_ch) const;
  torch::Tensor expand(torch::Tensor stats, torch::Tensor durs,
                       torch::Tensor y_lengths) const;

  // graph stages
  std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
  text_encoder(torch::Tensor ids, torch::Tensor lengths) const;
  torch::Tensor attention(torch::Tensor x, torch::Tensor attn_mask,
                          const std::string& mod) const;
  torch::Tensor wn(torch::Tensor x, torch::Tensor mask,
                   c10::optional<torch::Tensor> g, const std::string& mod,
                   long n_layers, long kernel, long dil_rate) const;
  torch::Tensor flow_reverse(torch::Tensor x, torch::Tensor mask,
                             c10::optional<torch::Tensor> g) const;
  torch::Tensor dds_conv(torch::Tensor x, torch::Tensor mask,
                         c10::optional<torch::Tensor> g,
                         const std::string& mod, long n_layers,
                         long kernel) const;
  torch::Tensor sdp_infer(torch::Tensor x, torch::Tensor mask,
                          c10::optional<torch::Tensor> g, double noise_w,
                          torch::Tensor noise) const;
  torch::Tensor generator(torch::Tensor z_masked,
                          c10::optional<torch::Tensor> g,
                          c10::optional<torch::Tensor> lengths) const;

  torch::Tensor masked_noise(long B, long C, long T_max,
                             torch::Tensor lengths,
                             std::vector<torch::Generator>& gens) const;
};

// rational-quadratic spline (Durkan et al.), exposed for tests
std::pair<torch::Tensor, torch::Tensor> rq_spline(
    torch::Tensor inputs, torch::Tensor uw, torch::Tensor uh,
    torch::Tensor ud, bool inverse, double tail_bound);

}  // namespace sonata
