// Python bindings for the CDNA4 kernel library (_tfmx_C).
// Kernel semantics are defined by transformer_amd/ops/reference.py; shapes
// and dtypes are validated here so kernels can assume clean inputs.
#include <torch/extension.h>

#include <vector>

// ---- implemented in the .hip translation units ----------------------------
torch::Tensor gemm_nt(torch::Tensor a, torch::Tensor w, torch::Tensor bias,
                      int64_t epilogue);
torch::Tensor transpose2d(torch::Tensor a);
torch::Tensor colsum(torch::Tensor a);
torch::Tensor relu_bwd(torch::Tensor dy, torch::Tensor y);
torch::Tensor smoke_add(torch::Tensor a, torch::Tensor b);

std::vector<torch::Tensor> ln_fwd(torch::Tensor x, torch::Tensor res,
                                  torch::Tensor gamma, torch::Tensor beta,
                                  double eps);
std::vector<torch::Tensor> ln_bwd(torch::Tensor dy, torch::Tensor s,
                                  torch::Tensor gamma, torch::Tensor mean,
                                  torch::Tensor rstd);

std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor kv_pad,
                                    bool causal, double scale);
std::vector<torch::Tensor> attn_bwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor o,
                                    torch::Tensor dout, torch::Tensor lse,
                                    torch::Tensor kv_pad, bool causal,
                                    double scale);

torch::Tensor embed_pe_fwd(torch::Tensor tokens, torch::Tensor weight,
                           torch::Tensor pe);
torch::Tensor embed_pe_bwd(torch::Tensor dy, torch::Tensor tokens,
                           int64_t vocab);

std::vector<torch::Tensor> dropout_fwd(torch::Tensor x, double p,
                                       int64_t seed);
torch::Tensor dropout_bwd(torch::Tensor dy, torch::Tensor mask, double p);

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor targets,
                                  double batch_size, double label_smoothing);
torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor targets,
                     torch::Tensor lse, double dloss, double batch_size,
                     double label_smoothing);

void adam_fused(torch::Tensor master, torch::Tensor m, torch::Tensor v,
                torch::Tensor grad, torch::Tensor param, double lr,
                double beta1, double beta2, double eps, int64_t step);

torch::Tensor argmax_lastdim(torch::Tensor logits);
std::vector<int64_t> accuracy(torch::Tensor logits, torch::Tensor targets);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gemm_nt", &gemm_nt, "C[M,N] = A[M,K] @ W[N,K]^T + bias (epilogue: 0=none 1=relu)");
  m.def("transpose2d", &transpose2d);
  m.def("colsum", &colsum);
  m.def("relu_bwd", &relu_bwd);
  m.def("smoke_add", &smoke_add);
  m.def("ln_fwd", &ln_fwd);
  m.def("ln_bwd", &ln_bwd);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
  m.def("embed_pe_fwd", &embed_pe_fwd);
  m.def("embed_pe_bwd", &embed_pe_bwd);
  m.def("dropout_fwd", &dropout_fwd);
  m.def("dropout_bwd", &dropout_bwd);
  m.def("ce_fwd", &ce_fwd);
  m.def("ce_bwd", &ce_bwd);
  m.def("adam_fused", &adam_fused);
  m.def("argmax_lastdim", &argmax_lastdim);
  m.def("accuracy", &accuracy);
}
