// Python bindings for the code_intelligence_amd gfx950 kernels.
#include <torch/extension.h>
#include <string>
#include <vector>

namespace ci {
void lstm_seq_forward_lib(at::Tensor xp, at::Tensor bias, at::Tensor h0,
                          at::Tensor c0, at::Tensor w_hh, at::Tensor hs,
                          at::Tensor cs, at::Tensor gates);
void lstm_seq_forward_fused(at::Tensor xp, at::Tensor bias, at::Tensor h0,
                            at::Tensor c0, at::Tensor w_hh, at::Tensor hs,
                            at::Tensor cs, at::Tensor gates);
void lstm_seq_forward_lib_fp8(at::Tensor xp, at::Tensor bias, at::Tensor h0,
                              at::Tensor c0, at::Tensor w8, at::Tensor wscale,
                              at::Tensor hs, at::Tensor cs, at::Tensor gates);
void quantize_e4m3(at::Tensor src, at::Tensor dst, at::Tensor scale);
void lstm_seq_forward_gemv(at::Tensor xp, at::Tensor bias, at::Tensor h0,
                           at::Tensor c0, at::Tensor w_hh, at::Tensor hs,
                           at::Tensor cs, at::Tensor gates);
long lstm_seq_forward_gemv_persistent(at::Tensor xp, at::Tensor bias,
                                      at::Tensor h0, at::Tensor c0,
                                      at::Tensor w_hh, at::Tensor hs,
                                      at::Tensor cs, at::Tensor gates,
                                      at::Tensor ws);
void lstm_seq_forward_gemv_fp8(at::Tensor xp, at::Tensor bias, at::Tensor h0,
                               at::Tensor c0, at::Tensor w8, at::Tensor wscale,
                               at::Tensor hs, at::Tensor cs, at::Tensor gates);
void lstm_seq_backward(at::Tensor dhs, at::Tensor dhT, at::Tensor dcT,
                       at::Tensor gates, at::Tensor hs, at::Tensor cs,
                       at::Tensor c0, at::Tensor w_hh, at::Tensor dgates,
                       at::Tensor dh0, at::Tensor dc0);
at::Tensor concat_pool(at::Tensor hidden, at::Tensor lengths);
std::vector<at::Tensor> qrnn_fo_pool_fwd(at::Tensor gates, at::Tensor c0);
std::vector<at::Tensor> qrnn_fo_pool_bwd(at::Tensor gates, at::Tensor c,
                                         at::Tensor c0, at::Tensor dh,
                                         at::Tensor dcT);
void ce_rowstats(at::Tensor logits, at::Tensor targets, at::Tensor bias,
                 at::Tensor lse, at::Tensor tgt);
void ce_dlogits(at::Tensor logits, at::Tensor targets, at::Tensor bias,
                at::Tensor lse, at::Tensor scale);
void ce_dlogits_dual(at::Tensor logits, at::Tensor targets, at::Tensor bias,
                     at::Tensor lse, at::Tensor scale, at::Tensor scratch8,
                     double store_scale);
at::Tensor artar_forward(at::Tensor out, at::Tensor r);
std::vector<at::Tensor> artar_backward(at::Tensor out, at::Tensor r,
                                       at::Tensor dloss, double ca, double cb);
void fused_adamw(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
                 std::vector<at::Tensor> masters, std::vector<at::Tensor> eas_,
                 std::vector<at::Tensor> eass, double lr, double b1, double b2,
                 double eps, double wd, double bc1, double bc2);
at::Tensor emb_gather(at::Tensor weight, at::Tensor ids, at::Tensor rowmask);
std::vector<std::string> tokenize_core(const std::string& text);
std::vector<std::vector<std::string>> tokenize_core_batch(
    const std::vector<std::string>& texts);
at::Tensor emb_scatter(at::Tensor gout, at::Tensor ids, at::Tensor rowmask,
                       long V, long pad_idx);
at::Tensor dropconnect_apply(at::Tensor w, long seed, double p);
void dropconnect_grad_(at::Tensor g, long seed, double p);
}  // namespace ci

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("lstm_seq_forward_lib", &ci::lstm_seq_forward_lib,
        "LSTM sequence forward (hipBLASLt GEMM + pointwise cell kernel)");
  m.def("lstm_seq_forward_fused", &ci::lstm_seq_forward_fused,
        "LSTM sequence forward (fused MFMA cell kernel)");
  m.def("lstm_seq_forward_lib_fp8", &ci::lstm_seq_forward_lib_fp8,
        "LSTM sequence forward (fp8 recurrent scaled_mm + pointwise cell)");
  m.def("quantize_e4m3", &ci::quantize_e4m3,
        "one-pass e4m3 quantize: dst = e4m3(src/scale)");
  m.def("lstm_seq_forward_gemv", &ci::lstm_seq_forward_gemv,
        "LSTM sequence forward (fused GEMV+cell kernel, small batch)");
  m.def("lstm_seq_forward_gemv_persistent",
        &ci::lstm_seq_forward_gemv_persistent,
        "whole-sequence persistent GEMV+cell kernel (grid barrier)");
  m.def("lstm_seq_forward_gemv_fp8", &ci::lstm_seq_forward_gemv_fp8,
        "LSTM sequence forward (fp8-weight GEMV+cell kernel)");
  m.def("lstm_seq_backward", &ci::lstm_seq_backward,
        "LSTM sequence backward (pointwise + per-step GEMM)");
  m.def("concat_pool", &ci::concat_pool, "masked mean/max/last concat pool");
  m.def("qrnn_fo_pool_fwd", &ci::qrnn_fo_pool_fwd,
        "QRNN fo-pool forward scan (activates gates in place)");
  m.def("qrnn_fo_pool_bwd", &ci::qrnn_fo_pool_bwd,
        "QRNN fo-pool backward scan (pre-activation gate grads)");
  m.def("ce_rowstats", &ci::ce_rowstats, "CE row logsumexp + target logit");
  m.def("ce_dlogits", &ci::ce_dlogits, "in-place (softmax-onehot)*scale");
  m.def("ce_dlogits_dual", &ci::ce_dlogits_dual,
        "in-place bf16 dlogits + e4m3 scratch copy for the fp8 dh GEMM");
  m.def("artar_forward", &ci::artar_forward,
        "fused AR/TAR partial sums (one pass over out and r)");
  m.def("artar_backward", &ci::artar_backward,
        "fused AR/TAR gradient kernel (dout and dr in one pass)");
  m.def("fused_adamw", &ci::fused_adamw, "fused AdamW step");
  m.def("emb_gather", &ci::emb_gather, "embedding gather with row dropout");
  m.def("tokenize_core", &ci::tokenize_core, "native ASCII tokenizer core");
  m.def("tokenize_core_batch", &ci::tokenize_core_batch,
        "native ASCII tokenizer core (batch, GIL released)");
  m.def("emb_scatter", &ci::emb_scatter, "embedding grad scatter-add");
  m.def("dropconnect_apply", &ci::dropconnect_apply,
        "seeded DropConnect mask-scale (no mask tensor)");
  m.def("dropconnect_grad_", &ci::dropconnect_grad_,
        "in-place seeded DropConnect grad mask");
}
