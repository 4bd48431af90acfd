// VitsEngine implementation — see vits_engine.h.
//
// The graph math mirrors sonata_amd/models/vits.py (the numerics oracle);
// every GPU hot op dispatches to the same HIP kernel entry points the
// Python path uses, so both runtimes share one kernel library.
#include "vits_engine.h"

#include <ATen/ATen.h>
#include <ATen/detail/CUDAHooksInterface.h>
#include <torch/types.h>

#include <cmath>
#include <fstream>
#include <sstream>

#include "minijson.h"

// HIP kernel entry points (host wrappers in csrc/*.hip) — same symbols
// the Python extension binds.
torch::Tensor layer_norm_ct(torch::Tensor x, c10::optional<torch::Tensor> res,
                            torch::Tensor gamma, torch::Tensor beta,
                            double eps);
torch::Tensor fused_gate(torch::Tensor x, c10::optional<torch::Tensor> g,
                         long n_channels);
torch::Tensor prior_sample(torch::Tensor m, torch::Tensor logs,
                           torch::Tensor mask, torch::Tensor noise,
                           double noise_scale);
torch::Tensor expand_states(torch::Tensor stats, torch::Tensor durs,
                            long F_max);
torch::Tensor seeded_noise(long B, long C, long T_max, torch::Tensor lens,
                           torch::Tensor seeds, torch::ScalarType dtype);
torch::Tensor row_ln_cl(torch::Tensor x, c10::optional<torch::Tensor> resid,
                        torch::Tensor gamma, torch::Tensor beta, double eps);
torch::Tensor conv1d_fused(torch::Tensor x, torch::Tensor w_perm,
                           c10::optional<torch::Tensor> bias, long Cout,
                           long k, long stride, long padding, long dilation,
                           long groups, double pre_lrelu, long act_mode,
                           double post_slope,
                           c10::optional<torch::Tensor> residual);
torch::Tensor conv1d_cl_fused(torch::Tensor x, torch::Tensor w_perm,
                              c10::optional<torch::Tensor> bias, long Cout,
                              long k, long padding, long dilation,
                              double pre_lrelu, long act_mode,
                              double post_slope,
                              c10::optional<torch::Tensor> residual,
                              c10::optional<torch::Tensor> out_lens);
torch::Tensor convtranspose1d_cl_fused(torch::Tensor x, torch::Tensor w_perm,
                                       c10::optional<torch::Tensor> bias,
                                       long Cout, long k, long stride,
                                       long padding, double pre_lrelu,
                                       c10::optional<torch::Tensor> out_lens);
torch::Tensor resblock_pair_cl_fused(torch::Tensor x, torch::Tensor w1_perm,
                                     torch::Tensor b1, torch::Tensor w2_perm,
                                     torch::Tensor b2, long k, long dil,
                                     c10::optional<torch::Tensor> out_lens,
                                     c10::optional<torch::Tensor> accum,
                                     double out_scale);
torch::Tensor resblock_chain_cl_fused(
    torch::Tensor x, torch::Tensor w_all, torch::Tensor b_all, long k,
    long d1, long d2, long d3, c10::optional<torch::Tensor> out_lens,
    c10::optional<torch::Tensor> accum, double out_scale);
torch::Tensor fused_gate_cl(torch::Tensor x, c10::optional<torch::Tensor> g,
                            long n_channels);
torch::Tensor attn_relpos_cl(torch::Tensor qkv, torch::Tensor rel_k,
                             torch::Tensor rel_v,
                             c10::optional<torch::Tensor> lens, long H,
                             long window, double scale);
torch::Tensor depthwise_cl(torch::Tensor x, torch::Tensor w,
                           c10::optional<torch::Tensor> bias, long dil,
                           long pad);

namespace sonata {

namespace {

constexpr double kLRelu = 0.1;

long round_up(long v, long m) { return (v + m - 1) / m * m; }

std::string read_file(const std::string& path) {
  std::ifstream f(path, std::ios::binary);
  TORCH_CHECK(f.good(), "cannot open ", path);
  std::stringstream ss;
  ss << f.rdbuf();
  return ss.str();
}

torch::Tensor sequence_mask(torch::Tensor lengths, long max_len) {
  auto pos = torch::arange(max_len, lengths.options());
  return (pos.unsqueeze(0) < lengths.unsqueeze(1))
      .unsqueeze(1)
      .to(torch::kFloat32);
}

// safetensors: u64 header length | JSON header | raw data
std::unordered_map<std::string, torch::Tensor> load_safetensors(
    const std::string& path) {
  std::string blob = read_file(path);
  TORCH_CHECK(blob.size() >= 8, "safetensors: truncated ", path);
  uint64_t hlen;
  memcpy(&hlen, blob.data(), 8);
  TORCH_CHECK(8 + hlen <= blob.size(), "safetensors: bad header length");
  auto header = minijson::parse(blob.substr(8, hlen));
  const char* base = blob.data() + 8 + hlen;
  const size_t data_size = blob.size() - 8 - hlen;
  std::unordered_map<std::string, torch::Tensor> out;
  for (auto& kv : header->obj) {
    if (kv.first == "__metadata__") continue;
    const minijson::Value& t = *kv.second;
    std::string dtype = t.str_at("dtype", "F32");
    torch::Dtype dt;
    long esize;
    if (dtype == "F32") { dt = torch::kFloat32; esize = 4; }
    else if (dtype == "F16") { dt = torch::kFloat16; esize = 2; }
    else if (dtype == "BF16") { dt = torch::kBFloat16; esize = 2; }
    else if (dtype == "I64") { dt = torch::kInt64; esize = 8; }
    else if (dtype == "I32") { dt = torch::kInt32; esize = 4; }
    else { TORCH_CHECK(false, "safetensors: dtype ", dtype); }
    std::vector<int64_t> shape;
    for (auto& d : t.get("shape")->arr) shape.push_back((int64_t)d->num);
    auto offs = t.get("data_offsets");
    size_t lo = (size_t)offs->arr[0]->num, hi = (size_t)offs->arr[1]->num;
    TORCH_CHECK(hi <= data_size && lo <= hi, "safetensors: bad offsets");
    long numel = 1;
    for (auto d : shape) numel *= d;
    TORCH_CHECK((size_t)(numel * esize) == hi - lo,
                "safetensors: size mismatch for ", kv.first);
    auto tensor = torch::from_blob((void*)(base + lo), shape,
                                   torch::TensorOptions().dtype(dt))
                      .clone();
    out[kv.first] = tensor;
  }
  return out;
}

}  // namespace

// ---------------------------------------------------------------------- //
// construction
// ---------------------------------------------------------------------- //
VitsEngine::VitsEngine(const std::string& config_path, torch::Device device,
                       torch::Dtype dtype)
    : device_(device), dtype_(dtype) {
  auto root = minijson::parse(read_file(config_path));
  const minijson::Value& d = *root;
  // quality presets (config.py QUALITY_PRESETS)
  std::string quality = "medium";
  if (auto* audio = d.get("audio")) {
    quality = audio->str_at("quality", "medium");
    cfg_.sample_rate = (long)audio->num_at(
        "sample_rate", quality == "x_low" || quality == "low" ? 16000 : 22050);
  }
  if (quality == "x_low") {
    cfg_.inter = 96; cfg_.hidden = 96; cfg_.filter = 384; cfg_.n_layers = 3;
    cfg_.up_init_ch = 256;
  }
  if (auto* arch = d.get("architecture")) {
    cfg_.inter = (long)arch->num_at("inter_channels", cfg_.inter);
    cfg_.hidden = (long)arch->num_at("hidden_channels", cfg_.hidden);
    cfg_.filter = (long)arch->num_at("filter_channels", cfg_.filter);
    cfg_.n_heads = (long)arch->num_at("n_heads", cfg_.n_heads);
    cfg_.n_layers = (long)arch->num_at("n_layers", cfg_.n_layers);
    cfg_.kernel_size = (long)arch->num_at("kernel_size", cfg_.kernel_size);
    cfg_.window_size = (long)arch->num_at("window_size", cfg_.window_size);
    cfg_.gin = (long)arch->num_at("gin_channels", cfg_.gin);
    cfg_.up_init_ch =
        (long)arch->num_at("upsample_initial_channel", cfg_.up_init_ch);
    auto read_longs = [&](const char* key, std::vector<long>& dst) {
      if (auto* v = arch->get(key)) {
        dst.clear();
        for (auto& e : v->arr) dst.push_back((long)e->num);
      }
    };
    read_longs("resblock_kernel_sizes", cfg_.resblock_ks);
    read_longs("upsample_rates", cfg_.up_rates);
    read_longs("upsample_kernel_sizes", cfg_.up_ks);
    if (auto* v = arch->get("resblock_dilation_sizes")) {
      cfg_.resblock_dil.clear();
      for (auto& row : v->arr) {
        std::vector<long> r;
        for (auto& e : row->arr) r.push_back((long)e->num);
        cfg_.resblock_dil.push_back(r);
      }
    }
  }
  cfg_.num_speakers = (long)d.num_at("num_speakers", 1);
  if (cfg_.num_speakers > 1 && cfg_.gin == 0) cfg_.gin = 256;
  cfg_.num_symbols = (long)d.num_at("num_symbols", 0);
  if (auto* inf = d.get("inference")) {
    cfg_.noise_scale = inf->num_at("noise_scale", 0.667);
    cfg_.length_scale = inf->num_at("length_scale", 1.0);
    cfg_.noise_w = inf->num_at("noise_w", 0.8);
  }
  if (auto* pm = d.get("phoneme_id_map")) {
    for (auto& kv : pm->obj) {
      std::vector<long> ids;
      for (auto& e : kv.second->arr) ids.push_back((long)e->num);
      cfg_.phoneme_id_map[kv.first] = ids;
    }
  }

  // weights: <stem>.safetensors next to the config
  std::string stem = config_path;
  auto strip = [&](const std::string& suffix) {
    if (stem.size() > suffix.size() &&
        stem.compare(stem.size() - suffix.size(), suffix.size(), suffix) == 0)
      stem = stem.substr(0, stem.size() - suffix.size());
  };
  strip(".json");
  strip(".onnx");
  P_ = load_safetensors(stem + ".safetensors");
  for (auto& kv : P_) kv.second = kv.second.to(device_, dtype_);
}

torch::Tensor VitsEngine::p(const std::string& name) const {
  auto it = P_.find(name);
  TORCH_CHECK(it != P_.end(), "missing weight: ", name);
  return it->second;
}

c10::optional<torch::Tensor> VitsEngine::maybe(const std::string& name) const {
  auto it = P_.find(name);
  if (it == P_.end()) return c10::nullopt;
  return it->second;
}

// [Cout,Cin,k] -> [k,CoutP,CinP] bf16 (functional.py _conv_weight_mfma)
torch::Tensor VitsEngine::perm_conv(const std::string& wname) const {
  auto it = cache_.find("perm:" + wname);
  if (it != cache_.end()) return it->second;
  auto w = p(wname);
  long Cout = w.size(0), Cin = w.size(1), k = w.size(2);
  long bm = Cout >= 128 ? 128 : (Cout >= 64 ? 64 : 32);
  long CoutP = round_up(Cout, bm), CinP = round_up(Cin, 32);
  auto pm = torch::zeros(
      {k, CoutP, CinP},
      torch::TensorOptions().dtype(torch::kBFloat16).device(w.device()));
  pm.index_put_({torch::indexing::Slice(),
                 torch::indexing::Slice(0, Cout),
                 torch::indexing::Slice(0, Cin)},
                w.detach().permute({2, 0, 1}).to(torch::kBFloat16));
  pm = pm.contiguous();
  cache_["perm:" + wname] = pm;
  return pm;
}

// [Cin,Cout,k] -> [s,kr,CoutP,CinP] (functional.py _convt_weight_mfma)
torch::Tensor VitsEngine::perm_convt(const std::string& wname,
                                     long stride) const {
  auto it = cache_.find("permt:" + wname);
  if (it != cache_.end()) return it->second;
  auto w = p(wname);
  long Cin = w.size(0), Cout = w.size(1), k = w.size(2);
  long kr_max = (k + stride - 1) / stride;
  long bm = Cout >= 128 ? 128 : (Cout >= 64 ? 64 : 32);
  long CoutP = round_up(Cout, bm), CinP = round_up(Cin, 32);
  auto pm = torch::zeros(
      {stride, kr_max, CoutP, CinP},
      torch::TensorOptions().dtype(torch::kBFloat16).device(w.device()));
  auto wb = w.detach().to(torch::kBFloat16);
  for (long r = 0; r < stride; ++r)
    for (long m = 0; m < (k - r + stride - 1) / stride; ++m)
      pm.index_put_({r, m, torch::indexing::Slice(0, Cout),
                     torch::indexing::Slice(0, Cin)},
                    wb.index({torch::indexing::Slice(),
                              torch::indexing::Slice(), r + stride * m})
                        .t());
  pm = pm.contiguous();
  cache_["permt:" + wname] = pm;
  return pm;
}

torch::Tensor VitsEngine::bias_f32(const std::string& bname) const {
  auto it = cache_.find("b32:" + bname);
  if (it != cache_.end()) return it->second;
  auto b = p(bname).detach().to(torch::kFloat32).contiguous();
  cache_["b32:" + bname] = b;
  return b;
}

// ---------------------------------------------------------------------- //
// op helpers
// ---------------------------------------------------------------------- //
torch::Tensor VitsEngine::conv(torch::Tensor x, const std::string& mod,
                               long stride, long pad, long dil, long groups,
                               double pre_lrelu, double post_lrelu) const {
  auto w = p(mod + ".weight");
  auto b = maybe(mod + ".bias");
  long Cout = w.size(0), k = w.size(2);
  if (gpu() && x.scalar_type() == torch::kBFloat16 && groups == 1 &&
      stride == 1 && (k - 1) * dil <= 64) {
    c10::optional<torch::Tensor> bias32;
    if (b.has_value()) bias32 = bias_f32(mod + ".bias");
    return conv1d_fused(x.contiguous(), perm_conv(mod + ".weight"), bias32,
                        Cout, k, stride, pad, dil, groups,
                        pre_lrelu > 0 ? pre_lrelu : -1.0,
                        post_lrelu > 0 ? 1 : 0, post_lrelu, c10::nullopt);
  }
  if (gpu() && groups > 1) {
    // grouped/depthwise (DDSConv): the fused host wrapper\'s naive path
    // beats MIOpen\'s im2col+GEMM at these tiny shapes
    c10::optional<torch::Tensor> bias32;
    if (b.has_value()) bias32 = bias_f32(mod + ".bias");
    auto y = conv1d_fused(x.contiguous(), w.contiguous(), bias32, Cout, k,
                          stride, pad, dil, groups,
                          pre_lrelu > 0 ? pre_lrelu : -1.0,
                          post_lrelu > 0 ? 1 : 0, post_lrelu, c10::nullopt);
    return y;
  }
  if (pre_lrelu > 0) x = torch::leaky_relu(x, pre_lrelu);
  auto y = torch::conv1d(x, w, b.has_value() ? *b : torch::Tensor(), stride,
                         pad, dil, groups);
  if (post_lrelu > 0) y = torch::leaky_relu(y, post_lrelu);
  return y;
}

torch::Tensor VitsEngine::layer_norm(torch::Tensor x, const std::string& mod,
                                     c10::optional<torch::Tensor> res) const {
  auto gamma = p(mod + ".gamma");
  auto beta = p(mod + ".beta");
  if (gpu()) {
    return layer_norm_ct(x.contiguous(),
                         res.has_value() ? c10::optional<torch::Tensor>(
                                               res->contiguous())
                                         : c10::nullopt,
                         bias_f32(mod + ".gamma"), bias_f32(mod + ".beta"),
                         1e-5);
  }
  if (res.has_value()) x = x + *res;
  auto mean = x.mean(1, true);
  auto var = x.var(1, false, true);
  auto xhat = (x - mean) * torch::rsqrt(var + 1e-5);
  return xhat * gamma.view({1, -1, 1}) + beta.view({1, -1, 1});
}

torch::Tensor VitsEngine::gate(torch::Tensor x, c10::optional<torch::Tensor> g,
                               long n_ch) const {
  if (gpu()) {
    return fused_gate(x.contiguous(),
                      g.has_value() ? c10::optional<torch::Tensor>(
                                          g->contiguous())
                                    : c10::nullopt,
                      n_ch);
  }
  if (g.has_value()) x = x + *g;
  auto a = x.narrow(1, 0, n_ch);
  auto b = x.narrow(1, n_ch, n_ch);
  return torch::tanh(a) * torch::sigmoid(b);
}

torch::Tensor VitsEngine::expand(torch::Tensor stats, torch::Tensor durs,
                                 torch::Tensor y_lengths) const {
  long F_max = y_lengths.max().item<long>();
  if (gpu()) {
    return expand_states(stats.contiguous(),
                         durs.to(torch::kInt32).contiguous(), F_max);
  }
  long B = stats.size(0), T = stats.size(2);
  auto out = torch::zeros({B, stats.size(1), F_max}, stats.options());
  for (long b = 0; b < B; ++b) {
    auto cum = torch::cumsum(durs[b], 0);
    auto frames = torch::arange(F_max, stats.options().dtype(torch::kLong));
    auto idx = torch::searchsorted(cum, frames, false, true)
                   .clamp_max(T - 1);
    long lb = y_lengths[b].item<long>();
    auto sel = stats[b].index_select(1, idx.narrow(0, 0, lb));
    out[b].narrow(1, 0, lb).copy_(sel);
  }
  return out;
}

// ---------------------------------------------------------------------- //
// text encoder (vits.py TextEncoder / RelativeAttention / FFN)
// ---------------------------------------------------------------------- //
namespace {

torch::Tensor rel_to_abs(torch::Tensor x) {
  long b = x.size(0), h = x.size(1), l = x.size(2);
  x = torch::constant_pad_nd(x, {0, 1});
  auto x_flat = x.reshape({b, h, l * 2 * l});
  x_flat = torch::constant_pad_nd(x_flat, {0, l - 1});
  return x_flat.reshape({b, h, l + 1, 2 * l - 1})
      .index({torch::indexing::Slice(), torch::indexing::Slice(),
              torch::indexing::Slice(0, l),
              torch::indexing::Slice(l - 1, torch::indexing::None)});
}

torch::Tensor abs_to_rel(torch::Tensor x) {
  long b = x.size(0), h = x.size(1), l = x.size(2);
  x = torch::constant_pad_nd(x, {0, l - 1});
  auto x_flat = x.reshape({b, h, l * l + l * (l - 1)});
  x_flat = torch::constant_pad_nd(x_flat, {l, 0});
  return x_flat.reshape({b, h, l, 2 * l})
      .index({torch::indexing::Slice(), torch::indexing::Slice(),
              torch::indexing::Slice(),
              torch::indexing::Slice(1, torch::indexing::None)});
}

torch::Tensor rel_embeddings(torch::Tensor emb, long length, long window) {
  long pad_len = std::max(length - (window + 1), 0L);
  long start = std::max((window + 1) - length, 0L);
  if (pad_len > 0)
    emb = torch::constant_pad_nd(emb, {0, 0, pad_len, pad_len});
  return emb.index({torch::indexing::Slice(),
                    torch::indexing::Slice(start, start + 2 * length - 1)});
}

}  // namespace

torch::Tensor VitsEngine::attention(torch::Tensor x, torch::Tensor attn_mask,
                                    const std::string& mod) const {
  long B = x.size(0), C = x.size(1), T = x.size(2);
  long H = cfg_.n_heads, D = C / H;
  auto q = conv(x, mod + ".conv_q").view({B, H, D, T}).transpose(2, 3);
  auto k = conv(x, mod + ".conv_k").view({B, H, D, T}).transpose(2, 3);
  auto v = conv(x, mod + ".conv_v").view({B, H, D, T}).transpose(2, 3);
  double scale = 1.0 / std::sqrt((double)D);
  auto scores = torch::matmul(q * scale, k.transpose(-2, -1));
  auto rel_k = rel_embeddings(p(mod + ".emb_rel_k"), T, cfg_.window_size);
  auto rel_logits =
      torch::matmul(q * scale, rel_k.unsqueeze(0).transpose(-2, -1));
  scores = scores + rel_to_abs(rel_logits);
  scores = scores.masked_fill(attn_mask == 0, -1e4);
  auto pr = torch::softmax(scores, -1);
  auto out = torch::matmul(pr, v);
  auto rel_w = abs_to_rel(pr);
  auto rel_v = rel_embeddings(p(mod + ".emb_rel_v"), T, cfg_.window_size);
  out = out + torch::matmul(rel_w, rel_v.unsqueeze(0));
  out = out.transpose(2, 3).contiguous().view({B, C, T});
  return conv(out, mod + ".conv_o");
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor>
VitsEngine::text_encoder(torch::Tensor ids, torch::Tensor lengths) const {
  if (gpu() && dtype_ == torch::kBFloat16) {
    // channel-last encoder (mirrors vits.py TextEncoder._forward_cl):
    // the embedding is already [B,T,H]; attention heads are a free view;
    // FFN runs on cl MFMA convs with fused ReLU + ragged-row masking.
    auto lin = [&](torch::Tensor t, const std::string& mod) {
      return torch::linear(t, p(mod + ".weight").squeeze(-1),
                           p(mod + ".bias"));
    };
    auto x = torch::embedding(p("enc_p.emb.weight"), ids) *
             std::sqrt((double)cfg_.hidden);  // [B,T,H]
    auto mask_cl = sequence_mask(lengths, ids.size(1))
                       .to(x.dtype()).transpose(1, 2);  // [B,T,1]
    x = x * mask_cl;
    const long H = cfg_.n_heads, D = cfg_.hidden / cfg_.n_heads;
    auto lens32 = lengths.to(torch::kInt32).contiguous();
    for (long i = 0; i < cfg_.n_layers; ++i) {
      std::string li = std::to_string(i);
      std::string am = "enc_p.attn_layers." + li;
      auto xm = x * mask_cl;
      long B = x.size(0), T = x.size(1), C = x.size(2);
      double scale = 1.0 / std::sqrt((double)D);
      // fused attention kernel (csrc/attention_cl.hip): one QKV GEMM +
      // one kernel replace the QK^T/softmax/PV matmul chain + the
      // [B,h,T,2T-1] rel pad/reshape traffic
      auto wit = cache_.find("qkvw:" + am);
      if (wit == cache_.end()) {
        auto wq = p(am + ".conv_q.weight").squeeze(-1);
        auto wk = p(am + ".conv_k.weight").squeeze(-1);
        auto wv = p(am + ".conv_v.weight").squeeze(-1);
        cache_["qkvw:" + am] = torch::cat({wq, wk, wv}).contiguous();
        cache_["qkvb:" + am] =
            torch::cat({p(am + ".conv_q.bias"), p(am + ".conv_k.bias"),
                        p(am + ".conv_v.bias")}).contiguous();
        cache_["relk:" + am] = p(am + ".emb_rel_k")[0].contiguous();
        cache_["relv:" + am] = p(am + ".emb_rel_v")[0].contiguous();
      }
      auto qkv = torch::linear(xm, cache_["qkvw:" + am],
                               cache_["qkvb:" + am]);
      auto outt = attn_relpos_cl(qkv, cache_["relk:" + am],
                                 cache_["relv:" + am], lens32, H,
                                 cfg_.window_size, scale);
      auto y = lin(outt, am + ".conv_o");
      x = row_ln_cl(x.contiguous(), y.contiguous(),
                    p("enc_p.norm1." + li + ".gamma"),
                    p("enc_p.norm1." + li + ".beta"), 1e-5);
      std::string f1 = "enc_p.ffn_layers." + li + ".conv1";
      std::string f2 = "enc_p.ffn_layers." + li + ".conv2";
      long pad = cfg_.kernel_size / 2;
      auto wf1 = p(f1 + ".weight");
      auto f = conv1d_cl_fused((x * mask_cl).contiguous(), perm_conv(
                                   f1 + ".weight"), bias_f32(f1 + ".bias"),
                               wf1.size(0), wf1.size(2), pad, 1, -1.0,
                               1 /*relu: lrelu slope 0*/, 0.0, c10::nullopt,
                               lens32);
      auto wf2 = p(f2 + ".weight");
      f = conv1d_cl_fused(f, perm_conv(f2 + ".weight"),
                          bias_f32(f2 + ".bias"), wf2.size(0), wf2.size(2),
                          pad, 1, -1.0, 0, 0.0, c10::nullopt, lens32);
      x = row_ln_cl(x.contiguous(), f.contiguous(),
                    p("enc_p.norm2." + li + ".gamma"),
                    p("enc_p.norm2." + li + ".beta"), 1e-5);
    }
    auto stats = (lin(x, "enc_p.proj") * mask_cl).transpose(1, 2);
    auto chunks = stats.chunk(2, 1);
    auto x_mask = mask_cl.transpose(1, 2);
    return {(x.transpose(1, 2) * x_mask).contiguous(),
            chunks[0].contiguous(), chunks[1].contiguous(), x_mask};
  }
  auto x = torch::embedding(p("enc_p.emb.weight"), ids) *
           std::sqrt((double)cfg_.hidden);
  x = x.transpose(1, 2).contiguous();  // [B, H, T]
  auto x_mask = sequence_mask(lengths, ids.size(1)).to(x.dtype());
  auto attn_mask =
      (x_mask.unsqueeze(2) * x_mask.unsqueeze(-1)).squeeze(1);
  x = x * x_mask;
  for (long i = 0; i < cfg_.n_layers; ++i) {
    std::string li = std::to_string(i);
    auto y = attention(x * x_mask, attn_mask.unsqueeze(1),
                       "enc_p.attn_layers." + li);
    x = layer_norm(x, "enc_p.norm1." + li, y);
    // FFN
    long pad = cfg_.kernel_size / 2;
    auto f = conv(x * x_mask, "enc_p.ffn_layers." + li + ".conv1", 1, pad);
    f = torch::relu(f);
    f = conv(f * x_mask, "enc_p.ffn_layers." + li + ".conv2", 1, pad);
    f = f * x_mask;
    x = layer_norm(x, "enc_p.norm2." + li, f);
  }
  auto stats = conv(x, "enc_p.proj") * x_mask;
  auto chunks = stats.chunk(2, 1);
  return {x, chunks[0], chunks[1], x_mask};
}

// ---------------------------------------------------------------------- //
// WaveNet + residual coupling flow (reverse)
// ---------------------------------------------------------------------- //
torch::Tensor VitsEngine::wn(torch::Tensor x, torch::Tensor mask,
                             c10::optional<torch::Tensor> g,
                             const std::string& mod, long n_layers,
                             long kernel, long dil_rate) const {
  auto output = torch::zeros_like(x);
  long hidden = x.size(1);
  c10::optional<torch::Tensor> g_all;
  if (g.has_value() && has(mod + ".cond_layer.weight"))
    g_all = conv(*g, mod + ".cond_layer");
  for (long i = 0; i < n_layers; ++i) {
    long dilation = 1;
    for (long j = 0; j < i; ++j) dilation *= dil_rate;
    long pad = (kernel - 1) * dilation / 2;
    auto x_in = conv(x, mod + ".in_layers." + std::to_string(i), 1, pad,
                     dilation);
    c10::optional<torch::Tensor> g_l;
    if (g_all.has_value())
      g_l = g_all->narrow(1, i * 2 * hidden, 2 * hidden);
    auto acts = gate(x_in, g_l, hidden);
    auto res_skip =
        conv(acts, mod + ".res_skip_layers." + std::to_string(i));
    if (i < n_layers - 1) {
      x = (x + res_skip.narrow(1, 0, hidden)) * mask;
      output = output + res_skip.narrow(1, hidden, hidden);
    } else {
      output = output + res_skip;
    }
  }
  return output * mask;
}

torch::Tensor VitsEngine::flow_reverse(torch::Tensor x, torch::Tensor mask,
                                       c10::optional<torch::Tensor> g) const {
  long half = cfg_.inter / 2;
  if (gpu() && x.scalar_type() == torch::kBFloat16) {
    // channel-last flow (mirrors vits.py reverse_cl): WN k-tap convs as
    // cl MFMA kernels, 1x1s as linears, gate via fused_gate_cl
    auto lin = [&](torch::Tensor t, const std::string& mod) {
      auto w = p(mod + ".weight").squeeze(-1);
      return torch::linear(t, w, p(mod + ".bias"));
    };
    auto xc = x.transpose(1, 2).contiguous();       // [B,F,C]
    auto mc = mask.transpose(1, 2).contiguous();    // [B,F,1]
    const long H = cfg_.hidden;
    for (long f = 3; f >= 0; --f) {
      xc = torch::flip(xc, {-1});
      std::string mod = "flow.flows." + std::to_string(f);
      auto x0 = xc.narrow(-1, 0, half).contiguous();
      auto x1 = xc.narrow(-1, half, half);
      auto h = lin(x0, mod + ".pre") * mc;
      // WN channel-last
      {
        auto output = torch::zeros_like(h);
        c10::optional<torch::Tensor> g_all;
        if (g.has_value() && has(mod + ".enc.cond_layer.weight"))
          g_all = torch::linear(
              g->squeeze(-1),
              p(mod + ".enc.cond_layer.weight").squeeze(-1),
              p(mod + ".enc.cond_layer.bias"));
        for (long i = 0; i < 4; ++i) {
          std::string li = std::to_string(i);
          auto cw = p(mod + ".enc.in_layers." + li + ".weight");
          auto x_in = conv1d_cl_fused(
              h.contiguous(), perm_conv(mod + ".enc.in_layers." + li +
                                        ".weight"),
              bias_f32(mod + ".enc.in_layers." + li + ".bias"),
              cw.size(0), cw.size(2), (cw.size(2) - 1) / 2, 1, -1.0, 0, 0.0,
              c10::nullopt, c10::nullopt);
          c10::optional<torch::Tensor> g_l;
          if (g_all.has_value())
            g_l = g_all->narrow(-1, i * 2 * H, 2 * H);
          auto acts = fused_gate_cl(x_in, g_l, H);
          auto res_skip = lin(acts, mod + ".enc.res_skip_layers." + li);
          if (i < 3) {
            h = (h + res_skip.narrow(-1, 0, H)) * mc;
            output = output + res_skip.narrow(-1, H, H);
          } else {
            output = output + res_skip;
          }
        }
        h = output * mc;
      }
      auto m = lin(h, mod + ".post") * mc;
      x1 = (x1 - m) * mc;
      xc = torch::cat({x0, x1}, -1);
    }
    return xc.transpose(1, 2).contiguous();
  }
  for (long f = 3; f >= 0; --f) {
    x = torch::flip(x, {1});
    std::string mod = "flow.flows." + std::to_string(f);
    auto x0 = x.narrow(1, 0, half);
    auto x1 = x.narrow(1, half, half);
    auto h = conv(x0, mod + ".pre") * mask;
    h = wn(h, mask, g, mod + ".enc", 4, 5, 1);
    auto m = conv(h, mod + ".post") * mask;
    x1 = (x1 - m) * mask;
    x = torch::cat({x0, x1}, 1);
  }
  return x;
}

// ---------------------------------------------------------------------- //
// stochastic duration predictor (reverse / sampling path)
// ---------------------------------------------------------------------- //
std::pair<torch::Tensor, torch::Tensor> rq_spline(
    torch::Tensor inputs, torch::Tensor uw, torch::Tensor uh,
    torch::Tensor ud, bool inverse, double tail_bound) {
  const double min_bin_width = 1e-3, min_bin_height = 1e-3,
               min_derivative = 1e-3;
  auto inside = (inputs >= -tail_bound) & (inputs <= tail_bound);
  auto outputs = torch::zeros_like(inputs);
  auto logabsdet = torch::zeros_like(inputs);
  outputs.masked_scatter_(~inside, inputs.masked_select(~inside));

  double constant = std::log(std::exp(1.0 - min_derivative) - 1.0);
  ud = torch::constant_pad_nd(ud, {1, 1}, constant);

  if (!inside.any().item<bool>()) return {outputs, logabsdet};

  long num_bins = uw.size(-1);
  auto sel = [&](torch::Tensor t) {
    return t.index({inside});
  };
  auto uw_i = sel(uw);
  auto uh_i = sel(uh);
  auto ud_i = sel(ud);
  auto x = inputs.masked_select(inside);

  auto widths = torch::softmax(uw_i, -1);
  widths = min_bin_width + (1 - min_bin_width * num_bins) * widths;
  auto cumwidths = torch::cumsum(widths, -1);
  cumwidths = torch::constant_pad_nd(cumwidths, {1, 0}, 0.0);
  cumwidths = (2 * tail_bound) * cumwidths - tail_bound;
  cumwidths.index_put_({torch::indexing::Ellipsis, 0}, -tail_bound);
  cumwidths.index_put_({torch::indexing::Ellipsis, -1}, tail_bound);
  widths = cumwidths.narrow(-1, 1, num_bins) -
           cumwidths.narrow(-1, 0, num_bins);

  auto derivatives = min_derivative + torch::softplus(ud_i);

  auto heights = torch::softmax(uh_i, -1);
  heights = min_bin_height + (1 - min_bin_height * num_bins) * heights;
  auto cumheights = torch::cumsum(heights, -1);
  cumheights = torch::constant_pad_nd(cumheights, {1, 0}, 0.0);
  cumheights = (2 * tail_bound) * cumheights - tail_bound;
  cumheights.index_put_({torch::indexing::Ellipsis, 0}, -tail_bound);
  cumheights.index_put_({torch::indexing::Ellipsis, -1}, tail_bound);
  heights = cumheights.narrow(-1, 1, num_bins) -
            cumheights.narrow(-1, 0, num_bins);

  torch::Tensor bin_idx;
  if (inverse)
    bin_idx = (torch::sum(x.unsqueeze(-1) >= cumheights, -1) - 1)
                  .unsqueeze(-1);
  else
    bin_idx = (torch::sum(x.unsqueeze(-1) >= cumwidths, -1) - 1)
                  .unsqueeze(-1);
  bin_idx = bin_idx.clamp(0, num_bins - 1);

  auto g1 = [&](torch::Tensor t) {
    return t.gather(-1, bin_idx).squeeze(-1);
  };
  auto in_cumwidths = g1(cumwidths);
  auto in_widths = g1(widths);
  auto in_cumheights = g1(cumheights);
  auto in_heights = g1(heights);
  auto delta = in_heights / in_widths;
  auto in_deriv = g1(derivatives);
  auto in_deriv_p1 = derivatives.narrow(-1, 1, derivatives.size(-1) - 1)
                         .gather(-1, bin_idx)
                         .squeeze(-1);

  torch::Tensor out, lad;
  if (inverse) {
    auto a = (x - in_cumheights) * (in_deriv + in_deriv_p1 - 2 * delta) +
             in_heights * (delta - in_deriv);
    auto bq = in_heights * in_deriv -
              (x - in_cumheights) * (in_deriv + in_deriv_p1 - 2 * delta);
    auto c = -delta * (x - in_cumheights);
    auto disc = (bq.pow(2) - 4 * a * c).clamp_min(0.0);
    auto root = (2 * c) / (-bq - torch::sqrt(disc));
    out = root * in_widths + in_cumwidths;
    auto tomt = root * (1 - root);
    auto denom = delta + (in_deriv + in_deriv_p1 - 2 * delta) * tomt;
    auto dn = delta.pow(2) * (in_deriv_p1 * root.pow(2) + 2 * delta * tomt +
                              in_deriv * (1 - root).pow(2));
    lad = -(torch::log(dn) - 2 * torch::log(denom));
  } else {
    auto theta = (x - in_cumwidths) / in_widths;
    auto tomt = theta * (1 - theta);
    auto numerator =
        in_heights * (delta * theta.pow(2) + in_deriv * tomt);
    auto denom = delta + (in_deriv + in_deriv_p1 - 2 * delta) * tomt;
    out = in_cumheights + numerator / denom;
    auto dn = delta.pow(2) * (in_deriv_p1 * theta.pow(2) + 2 * delta * tomt +
                              in_deriv * (1 - theta).pow(2));
    lad = torch::log(dn) - 2 * torch::log(denom);
  }
  outputs.masked_scatter_(inside, out);
  logabsdet.masked_scatter_(inside, lad);
  return {outputs, logabsdet};
}

torch::Tensor VitsEngine::dds_conv(torch::Tensor x, torch::Tensor mask,
                                   c10::optional<torch::Tensor> g,
                                   const std::string& mod, long n_layers,
                                   long kernel) const {
  if (g.has_value()) x = x + *g;
  long channels = x.size(1);
  for (long i = 0; i < n_layers; ++i) {
    long dilation = 1;
    for (long j = 0; j < i; ++j) dilation *= kernel;
    long pad = (kernel - 1) * dilation / 2;
    std::string li = std::to_string(i);
    auto y = conv(x * mask, mod + ".convs_sep." + li, 1, pad, dilation,
                  channels);
    y = layer_norm(y, mod + ".norms_1." + li);
    y = torch::gelu(y);
    y = conv(y, mod + ".convs_1x1." + li);
    y = layer_norm(y, mod + ".norms_2." + li);
    y = torch::gelu(y);
    x = x + y;
  }
  return x * mask;
}

torch::Tensor VitsEngine::sdp_infer(torch::Tensor x, torch::Tensor mask,
                                    c10::optional<torch::Tensor> g,
                                    double noise_w,
                                    torch::Tensor noise) const {
  if (gpu() && x.scalar_type() == torch::kBFloat16) {
    // channel-last SDP (mirrors vits.py _infer_cl): DDS/1x1 stages on
    // [B,T,C] rows, 2-channel flow state + spline stay channel-first
    auto lin = [&](torch::Tensor t, const std::string& mod) {
      return torch::linear(t, p(mod + ".weight").squeeze(-1),
                           p(mod + ".bias"));
    };
    auto dds_cl = [&](torch::Tensor t, torch::Tensor mc,
                      c10::optional<torch::Tensor> gc,
                      const std::string& mod) {
      if (gc.has_value()) t = t + *gc;
      const long C = t.size(-1);
      for (long i = 0; i < 3; ++i) {
        std::string li = std::to_string(i);
        long dilation = 1;
        for (long j = 0; j < i; ++j) dilation *= 3;
        long pad = (3 - 1) * dilation / 2;
        auto y = depthwise_cl((t * mc).contiguous(),
                              p(mod + ".convs_sep." + li + ".weight"),
                              bias_f32(mod + ".convs_sep." + li + ".bias"),
                              dilation, pad);
        y = row_ln_cl(y.contiguous(), c10::nullopt,
                      p(mod + ".norms_1." + li + ".gamma"),
                              p(mod + ".norms_1." + li + ".beta"), 1e-5);
        y = torch::gelu(y);
        y = lin(y, mod + ".convs_1x1." + li);
        y = row_ln_cl(y.contiguous(), c10::nullopt,
                      p(mod + ".norms_2." + li + ".gamma"),
                              p(mod + ".norms_2." + li + ".beta"), 1e-5);
        y = torch::gelu(y);
        t = t + y;
      }
      return t * mc;
    };
    auto mc = mask.transpose(1, 2).contiguous();  // [B,T,1]
    auto h = lin(x.detach().transpose(1, 2).contiguous(), "dp.pre");
    if (g.has_value() && has("dp.cond.weight"))
      h = h + lin(g->detach().squeeze(-1), "dp.cond").unsqueeze(1);
    h = dds_cl(h, mc, c10::nullopt, "dp.convs");
    h = lin(h, "dp.proj") * mc;

    auto z = noise * noise_w * mask;
    const long half = 1;
    auto conv_flow_cl = [&](torch::Tensor z, long idx) {
      std::string mod = "dp.flows." + std::to_string(idx);
      long B = z.size(0), T = z.size(2);
      auto z0 = z.narrow(1, 0, half);
      auto z1 = z.narrow(1, half, half);
      auto hz = lin(z0.transpose(1, 2).contiguous(), mod + ".pre");
      hz = dds_cl(hz, mc, h, mod + ".convs");
      hz = lin(hz, mod + ".proj") * mc;  // [B,T,half*(3b-1)]
      long num_bins = 10;
      auto h4 = hz.view({B, T, half, 3 * num_bins - 1})
                    .permute({0, 2, 1, 3});
      double scale = std::sqrt((double)192);
      auto uw = h4.index({torch::indexing::Ellipsis,
                          torch::indexing::Slice(0, num_bins)}) / scale;
      auto uh = h4.index({torch::indexing::Ellipsis,
                          torch::indexing::Slice(num_bins, 2 * num_bins)}) /
                scale;
      auto ud = h4.index({torch::indexing::Ellipsis,
                          torch::indexing::Slice(2 * num_bins,
                                                 torch::indexing::None)});
      auto res = rq_spline(z1, uw, uh, ud, true, 5.0);
      return torch::cat({z0, res.first}, 1) * mask;
    };
    auto flip = [&](torch::Tensor t) { return torch::flip(t, {1}); };
    z = flip(z);
    z = conv_flow_cl(z, 7);
    z = flip(z);
    z = conv_flow_cl(z, 5);
    z = flip(z);
    z = conv_flow_cl(z, 3);
    z = flip(z);
    auto m = p("dp.flows.0.m");
    auto logs = p("dp.flows.0.logs");
    z = (z - m) * torch::exp(-logs) * mask;
    return z.narrow(1, 0, 1);
  }
  x = conv(x.detach(), "dp.pre");
  if (g.has_value() && has("dp.cond.weight"))
    x = x + conv(g->detach(), "dp.cond");
  x = dds_conv(x, mask, c10::nullopt, "dp.convs", 3, 3);
  x = conv(x, "dp.proj") * mask;

  auto z = noise * noise_w * mask;
  // flows reversed: [EWA, CF0, Flip, CF1, Flip, CF2, Flip, CF3, Flip]
  // reversed -> [Flip, CF3, Flip, CF2, Flip, CF1, Flip, CF0, EWA];
  // python drops the final unused Flip pair: flows[:-2] + [flows[-1]]
  // => Flip, CF3, Flip, CF2, Flip, CF1, Flip, EWA? — mirror vits.py:
  // list(reversed(flows)) = [Flip,CF3,Flip,CF2,Flip,CF1,Flip,CF0,EWA],
  // flows[:-2]+[flows[-1]] = [Flip,CF3,Flip,CF2,Flip,CF1,Flip,EWA]...
  // NOTE: that drops CF0 and keeps EWA — replicate exactly.
  const long half = 1;  // channels=2, half=1
  auto conv_flow = [&](torch::Tensor z, long idx) {
    std::string mod = "dp.flows." + std::to_string(idx);
    auto z0 = z.narrow(1, 0, half);
    auto z1 = z.narrow(1, half, half);
    auto h = conv(z0, mod + ".pre");
    h = dds_conv(h, mask, x, mod + ".convs", 3, 3);
    h = conv(h, mod + ".proj") * mask;
    long B = z.size(0), T = z.size(2);
    long num_bins = 10;
    h = h.reshape({B, half, 3 * num_bins - 1, T}).permute({0, 1, 3, 2});
    double scale = std::sqrt((double)192);
    auto uw = h.index({torch::indexing::Ellipsis,
                       torch::indexing::Slice(0, num_bins)}) / scale;
    auto uh = h.index({torch::indexing::Ellipsis,
                       torch::indexing::Slice(num_bins, 2 * num_bins)}) /
              scale;
    auto ud = h.index({torch::indexing::Ellipsis,
                       torch::indexing::Slice(2 * num_bins,
                                              torch::indexing::None)});
    auto res = rq_spline(z1, uw, uh, ud, /*inverse=*/true, 5.0);
    z1 = res.first;
    return torch::cat({z0, z1}, 1) * mask;
  };
  // python module list: flows = [EWA(0), CF(1), Flip, CF(2), Flip, CF(3),
  // Flip, CF(4), Flip] — but state dict names: dp.flows.0 = EWA,
  // dp.flows.1/3/5/7 = ConvFlow, 2/4/6/8 = Flip (no params).
  // reversed+dropped sequence applied to z:
  //   Flip, CF(7), Flip, CF(5), Flip, CF(3), Flip, EWA(0)
  auto flip = [&](torch::Tensor z) { return torch::flip(z, {1}); };
  z = flip(z);
  z = conv_flow(z, 7);
  z = flip(z);
  z = conv_flow(z, 5);
  z = flip(z);
  z = conv_flow(z, 3);
  z = flip(z);
  // ElementwiseAffine reverse: (z - m) * exp(-logs) * mask
  auto m = p("dp.flows.0.m");
  auto logs = p("dp.flows.0.logs");
  z = (z - m) * torch::exp(-logs) * mask;
  return z.narrow(1, 0, 1);  // logw [B,1,T]
}

// ---------------------------------------------------------------------- //
// HiFi-GAN generator
// ---------------------------------------------------------------------- //
torch::Tensor VitsEngine::generator(torch::Tensor x,
                                    c10::optional<torch::Tensor> g,
                                    c10::optional<torch::Tensor> lengths)
    const {
  const long n_ups = (long)cfg_.up_rates.size();
  const long n_kernels = (long)cfg_.resblock_ks.size();
  c10::optional<torch::Tensor> lens = lengths;
  if (lens.has_value() && x.size(0) == 1) lens = c10::nullopt;

  if (gpu() && x.scalar_type() == torch::kBFloat16) {
    // channel-last serving path (vits.py Generator._forward_cl)
    c10::optional<torch::Tensor> lens32;
    auto to32 = [&](const c10::optional<torch::Tensor>& l)
        -> c10::optional<torch::Tensor> {
      if (!l.has_value()) return c10::nullopt;
      return l->to(torch::kInt32).contiguous();
    };
    auto xc = x.transpose(1, 2).contiguous();
    xc = conv1d_cl_fused(xc, perm_conv("dec.conv_pre.weight"),
                         bias_f32("dec.conv_pre.bias"),
                         p("dec.conv_pre.weight").size(0), 7, 3, 1, -1.0, 0,
                         0.0, c10::nullopt, to32(lens));
    if (g.has_value() && has("dec.cond.weight")) {
      xc = xc + conv(*g, "dec.cond").transpose(1, 2);
      if (lens.has_value()) {
        auto idx = torch::arange(xc.size(1), lens->options());
        xc = xc.masked_fill(
            (idx.unsqueeze(0) >= lens->unsqueeze(1)).unsqueeze(-1), 0);
      }
    }
    for (long i = 0; i < n_ups; ++i) {
      std::string ui = "dec.ups." + std::to_string(i);
      long s = cfg_.up_rates[i], k = cfg_.up_ks[i];
      if (lens.has_value()) lens = *lens * s;
      xc = convtranspose1d_cl_fused(
          xc, perm_convt(ui + ".weight", s), bias_f32(ui + ".bias"),
          p(ui + ".weight").size(1), k, s, (k - s) / 2, kLRelu, to32(lens));
      // fused resblock pairs; the MRF sum and /num_kernels fold into
      // the last pair\'s epilogue (same as the Python path)
      torch::Tensor xs;
      const long Cch = xc.size(2);
      static const bool chain_on = [] {  // opt-in (measured slower)
        const char* e = getenv("SONATA_RB_CHAIN");
        return e && e[0] == '1';
      }();
      for (long j = 0; j < n_kernels; ++j) {
        std::string rb =
            "dec.resblocks." + std::to_string(i * n_kernels + j);
        auto out = xc;
        long kk = cfg_.resblock_ks[j];
        const size_t npair = cfg_.resblock_dil[j].size();
        const bool last_rb = (j == n_kernels - 1);
        // whole-resblock chain kernel for the HBM-bound small-C stages
        long dsum = 3;
        for (long d : cfg_.resblock_dil[j]) dsum += d;
        const long S0 = 128 + (kk - 1) * dsum;
        if (chain_on && npair == 3 &&
            ((Cch == 32 && S0 <= 248) || (Cch == 64 && S0 <= 152))) {
          auto key = "chainw:" + rb;
          auto it = cache_.find(key);
          if (it == cache_.end()) {
            std::vector<torch::Tensor> ws, bs;
            for (size_t di = 0; di < 3; ++di) {
              std::string c1 = rb + ".convs1." + std::to_string(di);
              std::string c2 = rb + ".convs2." + std::to_string(di);
              ws.push_back(perm_conv(c1 + ".weight"));
              ws.push_back(perm_conv(c2 + ".weight"));
              bs.push_back(bias_f32(c1 + ".bias"));
              bs.push_back(bias_f32(c2 + ".bias"));
            }
            cache_[key] = torch::stack(ws).contiguous();
            cache_["chainb:" + rb] = torch::stack(bs).contiguous();
          }
          c10::optional<torch::Tensor> accum;
          if (xs.defined()) accum = xs;
          xs = resblock_chain_cl_fused(
              out, cache_[key], cache_["chainb:" + rb], kk,
              cfg_.resblock_dil[j][0], cfg_.resblock_dil[j][1],
              cfg_.resblock_dil[j][2], to32(lens), accum,
              last_rb ? 1.0 / (double)n_kernels : 1.0);
          continue;
        }
        for (size_t di = 0; di < npair; ++di) {
          long d = cfg_.resblock_dil[j][di];
          std::string c1 = rb + ".convs1." + std::to_string(di);
          std::string c2 = rb + ".convs2." + std::to_string(di);
          const bool last = (di == npair - 1);
          c10::optional<torch::Tensor> accum;
          if (last && xs.defined()) accum = xs;
          out = resblock_pair_cl_fused(
              out, perm_conv(c1 + ".weight"), bias_f32(c1 + ".bias"),
              perm_conv(c2 + ".weight"), bias_f32(c2 + ".bias"), kk, d,
              to32(lens), accum,
              (last && last_rb) ? 1.0 / (double)n_kernels : 1.0);
        }
        xs = out;
      }
      xc = xs;
    }
    xc = conv1d_cl_fused(xc, perm_conv("dec.conv_post.weight"), c10::nullopt,
                         1, 7, 3, 1, kLRelu, 2 /*tanh*/, 0.0, c10::nullopt,
                         to32(lens));
    return xc.transpose(1, 2);
  }

  // CPU / fp32 oracle path (channel-first)
  auto mask_rows = [&](torch::Tensor t,
                       const c10::optional<torch::Tensor>& l) {
    if (!l.has_value()) return t;
    auto idx = torch::arange(t.size(2), l->options());
    return t.masked_fill(
        (idx.unsqueeze(0) >= l->unsqueeze(1)).unsqueeze(1), 0);
  };
  x = conv(x, "dec.conv_pre", 1, 3);
  if (g.has_value() && has("dec.cond.weight")) x = x + conv(*g, "dec.cond");
  x = mask_rows(x, lens);
  for (long i = 0; i < n_ups; ++i) {
    std::string ui = "dec.ups." + std::to_string(i);
    long s = cfg_.up_rates[i], k = cfg_.up_ks[i];
    x = torch::conv_transpose1d(torch::leaky_relu(x, kLRelu),
                                p(ui + ".weight"), p(ui + ".bias"), s,
                                (k - s) / 2);
    if (lens.has_value()) {
      lens = *lens * s;
      x = mask_rows(x, lens);
    }
    torch::Tensor xs;
    const long n_kernels2 = n_kernels;
    for (long j = 0; j < n_kernels2; ++j) {
      std::string rb = "dec.resblocks." + std::to_string(i * n_kernels2 + j);
      auto out = x;
      long kk = cfg_.resblock_ks[j];
      for (size_t di = 0; di < cfg_.resblock_dil[j].size(); ++di) {
        long d = cfg_.resblock_dil[j][di];
        auto xt = conv(torch::Tensor(out), rb + ".convs1." +
                       std::to_string(di), 1, (kk - 1) * d / 2, d, 1, kLRelu);
        xt = mask_rows(xt, lens);
        out = conv(xt, rb + ".convs2." + std::to_string(di), 1, (kk - 1) / 2,
                   1, 1, kLRelu) + out;
        out = mask_rows(out, lens);
      }
      xs = xs.defined() ? xs + out : out;
    }
    x = xs / (double)n_kernels;
  }
  x = conv(x, "dec.conv_post", 1, 3, 1, 1, kLRelu);
  return torch::tanh(x);
}

// ---------------------------------------------------------------------- //
// full inference
// ---------------------------------------------------------------------- //
torch::Tensor VitsEngine::masked_noise(long B, long C, long T_max,
                                       torch::Tensor lengths,
                                       std::vector<torch::Generator>& gens)
    const {
  if (gpu()) {
    // ONE launch: counter-based normal noise keyed by (seed, c, t) —
    // replaces B per-row torch::randn launches + B .item() host syncs
    // per noise tensor (csrc/elementwise.hip seeded_noise_kernel; the
    // Python GPU path uses the same kernel, so engine == python holds)
    std::vector<long> sv;
    for (auto& g : gens) sv.push_back((long)g.current_seed());
    auto seeds = torch::from_blob(sv.data(), {(long)sv.size()},
                                  torch::kLong).clone().to(device_);
    auto lens32 = lengths.to(device_).to(torch::kInt);
    return seeded_noise(B, C, T_max, lens32.contiguous(),
                        seeds.contiguous(),
                        dtype_ == torch::kBFloat16 ? at::kBFloat16
                                                   : at::kFloat);
  }
  // CPU: per-utterance torch generators (the fp32 numerics oracle)
  auto opts = torch::TensorOptions().device(device_).dtype(dtype_);
  auto out = torch::zeros({B, C, T_max}, opts);
  auto lens_cpu = lengths.to(torch::kCPU);
  for (long b = 0; b < B; ++b) {
    long lb = lens_cpu[b].item<long>();
    if (lb <= 0) continue;
    out[b].narrow(1, 0, lb) = torch::randn({C, lb}, gens[b], opts);
  }
  return out;
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor>
VitsEngine::infer_encoder(torch::Tensor ids, torch::Tensor lengths,
                          c10::optional<torch::Tensor> sid,
                          double noise_scale, double length_scale,
                          double noise_w, const std::vector<int64_t>& seeds) {
  torch::NoGradGuard ng;
  ids = ids.to(device_);
  lengths = lengths.to(device_);
  long B = ids.size(0);
  std::vector<torch::Generator> gens;
  for (long b = 0; b < B; ++b) {
    auto g = device_.is_cuda()
                 ? at::detail::getCUDAHooks().getNewGenerator(device_.index())
                 : at::detail::createCPUGenerator();
    g.set_current_seed(b < (long)seeds.size() ? seeds[b] : 1234 + b);
    gens.push_back(g);
  }

  auto [x, m_p, logs_p, x_mask] = text_encoder(ids, lengths);
  c10::optional<torch::Tensor> g;
  if (has("emb_g.weight")) {
    torch::Tensor s = sid.has_value()
                          ? sid->to(device_)
                          : torch::zeros({B}, torch::TensorOptions()
                                                  .dtype(torch::kLong)
                                                  .device(device_));
    g = torch::embedding(p("emb_g.weight"), s).unsqueeze(-1);
  }
  auto sdp_noise = masked_noise(B, 2, ids.size(1), lengths, gens);
  auto logw = sdp_infer(x, x_mask, g, noise_w, sdp_noise);
  auto w = torch::exp(logw) * x_mask * length_scale;
  auto w_ceil = torch::ceil(w);
  auto y_lengths =
      torch::clamp_min(torch::sum(w_ceil, {1, 2}), 1).to(torch::kLong);
  long F_max = y_lengths.max().item<long>();
  auto y_mask = sequence_mask(y_lengths, F_max).to(x.dtype());
  auto durations = w_ceil.squeeze(1).to(torch::kLong);
  auto m_p_f = expand(m_p, durations, y_lengths);
  auto logs_p_f = expand(logs_p, durations, y_lengths);
  auto pnoise = masked_noise(B, m_p_f.size(1), m_p_f.size(2), y_lengths,
                             gens);
  torch::Tensor z_p;
  if (gpu()) {
    z_p = prior_sample(m_p_f.contiguous(), logs_p_f.contiguous(),
                       y_mask.contiguous(), pnoise, noise_scale);
  } else {
    z_p = (m_p_f + pnoise * torch::exp(logs_p_f) * noise_scale) * y_mask;
  }
  auto z = flow_reverse(z_p, y_mask, g);
  return {z, y_mask, g.has_value() ? *g : torch::Tensor()};
}

torch::Tensor VitsEngine::decode(torch::Tensor z, torch::Tensor y_mask,
                                 c10::optional<torch::Tensor> g,
                                 c10::optional<torch::Tensor> lengths) {
  torch::NoGradGuard ng;
  return generator(z * y_mask, g, lengths);
}

std::pair<torch::Tensor, torch::Tensor> VitsEngine::infer(
    torch::Tensor ids, torch::Tensor lengths,
    c10::optional<torch::Tensor> sid, double noise_scale,
    double length_scale, double noise_w, const std::vector<int64_t>& seeds) {
  auto [z, y_mask, g] = infer_encoder(ids, lengths, sid, noise_scale,
                                      length_scale, noise_w, seeds);
  auto y_lengths = y_mask.squeeze(1).sum(-1).to(torch::kLong);
  c10::optional<torch::Tensor> gopt;
  if (g.defined()) gopt = g;
  auto audio = decode(z, y_mask, gopt, y_lengths);
  return {audio, y_lengths * cfg_.hop()};
}

// ---------------------------------------------------------------------- //
// phoneme-id encoding (piper lib.rs:232-250: BOS ^=1, EOS $=2, PAD _=0
// interleaved after every phoneme)
// ---------------------------------------------------------------------- //
std::vector<int64_t> VitsEngine::phonemes_to_ids(
    const std::string& utf8) const {
  const auto& pm = cfg_.phoneme_id_map;
  auto ids_of = [&](const std::string& ch) -> const std::vector<long>* {
    auto it = pm.find(ch);
    return it == pm.end() ? nullptr : &it->second;
  };
  std::vector<int64_t> out;
  const std::vector<long>* bos = ids_of("^");
  const std::vector<long>* pad = ids_of("_");
  long bos_id = bos && !bos->empty() ? (*bos)[0] : 1;
  long eos_id = 2;
  if (const auto* e = ids_of("$"))
    if (!e->empty()) eos_id = (*e)[0];
  long pad_id = pad && !pad->empty() ? (*pad)[0] : 0;
  out.push_back(bos_id);
  // iterate utf-8 codepoints
  size_t i = 0;
  while (i < utf8.size()) {
    unsigned char c = utf8[i];
    size_t n = c < 0x80 ? 1 : (c < 0xE0 ? 2 : (c < 0xF0 ? 3 : 4));
    std::string ch = utf8.substr(i, n);
    i += n;
    if (const auto* ids = ids_of(ch)) {
      for (long v : *ids) out.push_back(v);
      out.push_back(pad_id);
    }
  }
  out.push_back(eos_id);
  return out;
}

}  // namespace sonata
