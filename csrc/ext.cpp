// Python bindings for the sonata_amd CDNA4 kernel library + the C++
// VitsEngine runtime.
#include <torch/extension.h>

#include "engine/vits_engine.h"

// elementwise.hip
torch::Tensor layer_norm_ct(torch::Tensor x, c10::optional<torch::Tensor> res,
                            torch::Tensor gamma, torch::Tensor beta,
                            double eps);
torch::Tensor fused_gate(torch::Tensor x, c10::optional<torch::Tensor> g,
                         long n_channels);
torch::Tensor fused_gate_cl(torch::Tensor x, c10::optional<torch::Tensor> g,
                            long n_channels);
torch::Tensor prior_sample(torch::Tensor m, torch::Tensor logs,
                           torch::Tensor mask, torch::Tensor noise,
                           double noise_scale);
torch::Tensor expand_states(torch::Tensor stats, torch::Tensor durs,
                            long F_max);
torch::Tensor mask_tail_(torch::Tensor x, torch::Tensor lens);
torch::Tensor seeded_noise(long B, long C, long T_max, torch::Tensor lens,
                           torch::Tensor seeds, torch::ScalarType dtype);
torch::Tensor row_ln_cl(torch::Tensor x, c10::optional<torch::Tensor> resid,
                        torch::Tensor gamma, torch::Tensor beta, double eps);
torch::Tensor depthwise_cl(torch::Tensor x, torch::Tensor w,
                           c10::optional<torch::Tensor> bias, long dil,
                           long pad);
// conv1d.hip
torch::Tensor conv1d_fused(torch::Tensor x, torch::Tensor w_perm,
                           c10::optional<torch::Tensor> bias, long Cout,
                           long k, long stride, long padding, long dilation,
                           long groups, double pre_lrelu, long act_mode,
                           double post_slope,
                           c10::optional<torch::Tensor> residual);
torch::Tensor convtranspose1d_fused(torch::Tensor x, torch::Tensor w_perm,
                                    c10::optional<torch::Tensor> bias,
                                    long Cout, long k, long stride,
                                    long padding, double pre_lrelu);
// conv1d_cl.hip (channel-last)
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
torch::Tensor conv1d_cl_wdirect(torch::Tensor x, torch::Tensor w_perm,
                                c10::optional<torch::Tensor> bias, long Cout,
                                long k, long padding, long dilation,
                                double pre_lrelu, long act_mode,
                                double post_slope,
                                c10::optional<torch::Tensor> residual,
                                c10::optional<torch::Tensor> out_lens);
torch::Tensor conv1d_direct_cl(torch::Tensor x, torch::Tensor w_perm,
                               c10::optional<torch::Tensor> bias, long Cout,
                               long k, long padding, long dilation,
                               double pre_lrelu, long act_mode,
                               double post_slope,
                               c10::optional<torch::Tensor> residual,
                               c10::optional<torch::Tensor> out_lens);
// attention_cl.hip
torch::Tensor attn_relpos_cl(torch::Tensor qkv, torch::Tensor rel_k,
                             torch::Tensor rel_v,
                             c10::optional<torch::Tensor> lens, long H,
                             long window, double scale);
// resblock_cl.hip
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

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "sonata_amd hand-written CDNA4 (gfx950) kernels";
  m.def("layer_norm_ct", &layer_norm_ct, "LayerNorm over channels of [B,C,T]");
  m.def("fused_gate", &fused_gate, "WaveNet tanh*sigmoid gate");
  m.def("fused_gate_cl", &fused_gate_cl, "channel-last WaveNet gate");
  m.def("prior_sample", &prior_sample, "z=(m+eps*exp(logs)*ns)*mask");
  m.def("expand_states", &expand_states, "duration length-regulator gather");
  m.def("mask_tail_", &mask_tail_, "in-place zero of x[b,:,lens[b]:]");
  m.def("seeded_noise",
        [](long B, long C, long T, torch::Tensor lens, torch::Tensor seeds,
           const std::string& dt) {
          return seeded_noise(B, C, T, lens, seeds,
                              dt == "bf16" ? at::kBFloat16 : at::kFloat);
        },
        "per-utterance counter-based normal noise, one launch");
  m.def("row_ln_cl", &row_ln_cl,
        "LayerNorm over channel-last rows with fused residual add");
  m.def("depthwise_cl", &depthwise_cl, "channel-last depthwise conv1d");
  m.def("conv1d_fused", &conv1d_fused, "MFMA conv1d with fused activations");
  m.def("convtranspose1d_fused", &convtranspose1d_fused,
        "MFMA transposed conv1d (phase-decomposed GEMMs)");
  m.def("conv1d_cl_fused", &conv1d_cl_fused,
        "channel-last MFMA conv1d, fused act/residual/mask");
  m.def("convtranspose1d_cl_fused", &convtranspose1d_cl_fused,
        "channel-last MFMA transposed conv1d, phase-merged");
  m.def("conv1d_cl_wdirect", &conv1d_cl_wdirect,
        "hybrid: A staged in LDS, weights direct from L2 (no W barriers)");
  m.def("conv1d_direct_cl", &conv1d_direct_cl,
        "zero-LDS direct channel-last conv (L1/L2-fed MFMA)");
  m.def("resblock_pair_cl_fused", &resblock_pair_cl_fused,
        "fused HiFi-GAN resblock conv pair (xt stays in LDS)");
  m.def("resblock_chain_cl_fused", &resblock_chain_cl_fused,
        "whole resblock (3 pairs) fused: intermediates never touch HBM");
  m.def("attn_relpos_cl", &attn_relpos_cl,
        "fused relative-position attention (QKT+band+softmax+PV+rel_v)",
        py::arg("qkv"), py::arg("rel_k"), py::arg("rel_v"), py::arg("lens"),
        py::arg("heads"), py::arg("window"), py::arg("scale"));

  // C++ inference runtime (csrc/engine): the ort-replacement executor.
  py::class_<sonata::VitsEngine>(m, "VitsEngine")
      .def(py::init([](const std::string& path, const std::string& device,
                       const std::string& dtype) {
             torch::Device dev(device);
             torch::Dtype dt =
                 dtype == "bf16" ? torch::kBFloat16 : torch::kFloat32;
             return new sonata::VitsEngine(path, dev, dt);
           }),
           py::arg("config_path"), py::arg("device") = "cpu",
           py::arg("dtype") = "f32")
      .def("infer",
           [](sonata::VitsEngine& e, torch::Tensor ids,
              torch::Tensor lengths, c10::optional<torch::Tensor> sid,
              double ns, double ls, double nw,
              std::vector<int64_t> seeds) {
             std::pair<torch::Tensor, torch::Tensor> r;
             {
               // release the GIL for the whole graph: concurrent server
               // handler threads serialize WAVs while the GPU runs
               py::gil_scoped_release rel;
               r = e.infer(ids, lengths, sid, ns, ls, nw, seeds);
             }
             return py::make_tuple(r.first, r.second);
           },
           py::arg("ids"), py::arg("lengths"),
           py::arg("sid") = py::none(), py::arg("noise_scale") = 0.667,
           py::arg("length_scale") = 1.0, py::arg("noise_w") = 0.8,
           py::arg("seeds") = std::vector<int64_t>{})
      .def("infer_encoder",
           [](sonata::VitsEngine& e, torch::Tensor ids,
              torch::Tensor lengths, c10::optional<torch::Tensor> sid,
              double ns, double ls, double nw,
              std::vector<int64_t> seeds) {
             std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> r;
             {
               py::gil_scoped_release rel;
               r = e.infer_encoder(ids, lengths, sid, ns, ls, nw, seeds);
             }
             return py::make_tuple(std::get<0>(r), std::get<1>(r),
                                   std::get<2>(r));
           },
           py::arg("ids"), py::arg("lengths"),
           py::arg("sid") = py::none(), py::arg("noise_scale") = 0.667,
           py::arg("length_scale") = 1.0, py::arg("noise_w") = 0.8,
           py::arg("seeds") = std::vector<int64_t>{})
      .def("decode",
           [](sonata::VitsEngine& e, torch::Tensor z, torch::Tensor y_mask,
              c10::optional<torch::Tensor> g,
              c10::optional<torch::Tensor> lengths) {
             py::gil_scoped_release rel;
             return e.decode(z, y_mask, g, lengths);
           },
           py::arg("z"), py::arg("y_mask"), py::arg("g") = py::none(),
           py::arg("lengths") = py::none())
      .def("phonemes_to_ids", &sonata::VitsEngine::phonemes_to_ids)
      .def_property_readonly(
          "sample_rate",
          [](sonata::VitsEngine& e) { return e.config().sample_rate; })
      .def_property_readonly(
          "num_speakers",
          [](sonata::VitsEngine& e) { return e.config().num_speakers; })
      .def_property_readonly("hop", [](sonata::VitsEngine& e) {
        return e.config().hop();
      });
}
