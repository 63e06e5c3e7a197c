// Python bindings for the CDNA4 kernel library (_tfmx_C).
// Kernel semantics are defined by transformer_amd/ops/reference.py; shapes
// and dtypes are validated here so kernels can assume clean inputs.
#include <torch/extension.h>

#include <vector>

// ---- implemented in the .hip translation units ----------------------------
torch::Tensor gemm_nt(torch::Tensor a, torch::Tensor w, torch::Tensor bias,
                      int64_t epilogue, c10::optional<torch::Tensor> out);
torch::Tensor gemm256_nt(torch::Tensor a, torch::Tensor w, torch::Tensor bias,
                         int64_t epilogue, c10::optional<torch::Tensor> out);
torch::Tensor gemm128_nt(torch::Tensor a, torch::Tensor w, torch::Tensor bias,
                         int64_t epilogue, c10::optional<torch::Tensor> out);
torch::Tensor gemm_dw(torch::Tensor dy, torch::Tensor x,
                      c10::optional<torch::Tensor> out,
                      c10::optional<torch::Tensor> db,
                      c10::optional<torch::Tensor> cw_cached);
torch::Tensor gemm_nn(torch::Tensor a, torch::Tensor b,
                      c10::optional<torch::Tensor> out);
torch::Tensor gemm_tn(torch::Tensor a, torch::Tensor b,
                      c10::optional<torch::Tensor> out);
torch::Tensor transpose2d(torch::Tensor a);
void transpose2d_into(torch::Tensor a, torch::Tensor out);
void transpose_batch(torch::Tensor desc);
torch::Tensor colsum(torch::Tensor a, c10::optional<torch::Tensor> out);
torch::Tensor gemm_uni_nt(torch::Tensor a, torch::Tensor w,
                          c10::optional<torch::Tensor> bias, int64_t epilogue,
                          c10::optional<torch::Tensor> out);
torch::Tensor gemm_uni_nn(torch::Tensor a, torch::Tensor w,
                          c10::optional<torch::Tensor> out);
torch::Tensor gemm_uni_tn(torch::Tensor dy, torch::Tensor x,
                          c10::optional<torch::Tensor> out, int64_t splitr);
bool gemm_uni_viable(int M, int N, int K);
torch::Tensor gemm_uni_nt_ab(torch::Tensor a, torch::Tensor w,
                             int64_t sched);
torch::Tensor relu_bwd(torch::Tensor dy, torch::Tensor y);
std::vector<torch::Tensor> relu_bwd_db(torch::Tensor dy, torch::Tensor y,
                                       c10::optional<torch::Tensor> db_out);
torch::Tensor smoke_add(torch::Tensor a, torch::Tensor b);

std::vector<torch::Tensor> ln_fwd(torch::Tensor x, torch::Tensor res,
                                  torch::Tensor gamma, torch::Tensor beta,
                                  double eps, double p, int64_t seed,
                                  c10::optional<torch::Tensor> seed_t);
std::vector<torch::Tensor> ln_bwd(torch::Tensor dy, torch::Tensor s,
                                  torch::Tensor gamma, torch::Tensor mean,
                                  torch::Tensor rstd,
                                  c10::optional<torch::Tensor> dgamma_out,
                                  c10::optional<torch::Tensor> dbeta_out,
                                  c10::optional<torch::Tensor> mask,
                                  double p);

std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor kv_pad,
                                    bool causal, double scale, int64_t trv);
std::vector<torch::Tensor> attn_bwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor o,
                                    torch::Tensor dout, torch::Tensor lse,
                                    torch::Tensor kv_pad, bool causal,
                                    double scale, int64_t mode);

torch::Tensor embed_pe_fwd(torch::Tensor tokens, torch::Tensor weight,
                           torch::Tensor pe);
torch::Tensor embed_pe_bwd(torch::Tensor dy, torch::Tensor tokens,
                           int64_t vocab,
                           c10::optional<torch::Tensor> out);

std::vector<torch::Tensor> dropout_fwd(torch::Tensor x, double p,
                                        int64_t seed,
                                        c10::optional<torch::Tensor> seed_t);
torch::Tensor dropout_bwd(torch::Tensor dy, torch::Tensor mask, double p);

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor targets,
                                  double batch_size, double label_smoothing);
torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor targets,
                     torch::Tensor lse, torch::Tensor dloss,
                     double batch_size, double label_smoothing);
std::vector<torch::Tensor> ce_fused(torch::Tensor logits,
                                    torch::Tensor targets,
                                    double batch_size,
                                    double label_smoothing);
void ce_scale(torch::Tensor dlogits, torch::Tensor dloss);

void adam_fused_dev(torch::Tensor master, torch::Tensor m, torch::Tensor v,
                    torch::Tensor grad, torch::Tensor param,
                    torch::Tensor step, torch::Tensor coefs, double d_model,
                    double warmup, double beta1, double beta2, double eps);
void adam_fused(torch::Tensor master, torch::Tensor m, torch::Tensor v,
                torch::Tensor grad, torch::Tensor param, double lr,
                double beta1, double beta2, double eps, int64_t step);

torch::Tensor argmax_lastdim(torch::Tensor logits);
std::vector<int64_t> accuracy(torch::Tensor logits, torch::Tensor targets);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gemm_nt", &gemm_nt,
        "C[M,N] = A[M,K] @ W[N,K]^T + bias (epilogue: 0=none 1=relu)",
        pybind11::arg("a"), pybind11::arg("w"), pybind11::arg("bias"),
        pybind11::arg("epilogue"), pybind11::arg("out") = pybind11::none());
  m.def("gemm_dw", &gemm_dw,
        "C[N,K] = dY[M,N]^T @ X[M,K] — deep-contraction dW via tr16 reads; "
        "optional fused bias grad db[N]",
        pybind11::arg("dy"), pybind11::arg("x"),
        pybind11::arg("out") = pybind11::none(),
        pybind11::arg("db") = pybind11::none(),
        pybind11::arg("cw") = pybind11::none());
  m.def("gemm128_nt", &gemm128_nt, "128x128 NT path, no 256 dispatch (A/B)",
        pybind11::arg("a"), pybind11::arg("w"), pybind11::arg("bias"),
        pybind11::arg("epilogue"), pybind11::arg("out") = pybind11::none());
  m.def("gemm256_nt", &gemm256_nt,
        "256x256 deep-pipelined NT GEMM (K%32==0; M/N any, edge-clamped)",
        pybind11::arg("a"), pybind11::arg("w"), pybind11::arg("bias"),
        pybind11::arg("epilogue"), pybind11::arg("out") = pybind11::none());
  m.def("gemm_nn", &gemm_nn, "C[M,N] = A[M,K] @ B[K,N] (dX)",
        pybind11::arg("a"), pybind11::arg("b"),
        pybind11::arg("out") = pybind11::none());
  m.def("gemm_tn", &gemm_tn, "C[M,N] = A[K,M]^T @ B[K,N] (dW)",
        pybind11::arg("a"), pybind11::arg("b"),
        pybind11::arg("out") = pybind11::none());
  m.def("transpose2d", &transpose2d);
  m.def("transpose2d_into", &transpose2d_into,
        "transpose a into out (out rows may be padded past a's row count)");
  m.def("transpose_batch", &transpose_batch,
        "batched 64x64-tile transposes from an int64 descriptor table");
  m.def("gemm_uni_nt", &gemm_uni_nt,
        "deep-pipelined 256-tile NT fwd GEMM (+bias/ReLU)",
        pybind11::arg("a"), pybind11::arg("w"),
        pybind11::arg("bias") = c10::nullopt, pybind11::arg("epilogue") = 0,
        pybind11::arg("out") = c10::nullopt);
  m.def("gemm_uni_nn", &gemm_uni_nn,
        "dX = dY @ W, W read red-major via tr16 (no weight transpose)",
        pybind11::arg("a"), pybind11::arg("w"),
        pybind11::arg("out") = c10::nullopt);
  m.def("gemm_uni_tn", &gemm_uni_tn,
        "dW = dY^T @ X, both operands tr16-read; optional split contraction",
        pybind11::arg("dy"), pybind11::arg("x"),
        pybind11::arg("out") = c10::nullopt, pybind11::arg("splitr") = 1);
  m.def("gemm_uni_viable", &gemm_uni_viable);
  m.def("gemm_uni_nt_ab", &gemm_uni_nt_ab,
        "schedule A/B variants of the uni NT kernel (bench only)");
  m.def("colsum", &colsum, pybind11::arg("a"),
        pybind11::arg("out") = pybind11::none());
  m.def("relu_bwd", &relu_bwd);
  m.def("relu_bwd_db", &relu_bwd_db,
        "dz = dy*(y>0) with the bias grad db = colsum(dz) fused",
        pybind11::arg("dy"), pybind11::arg("y"),
        pybind11::arg("db") = pybind11::none());
  m.def("smoke_add", &smoke_add);
  m.def("ln_fwd", &ln_fwd, pybind11::arg("x"), pybind11::arg("res"),
        pybind11::arg("gamma"), pybind11::arg("beta"), pybind11::arg("eps"),
        pybind11::arg("p") = 0.0, pybind11::arg("seed") = 0,
        pybind11::arg("seed_t") = pybind11::none());
  m.def("ln_bwd", &ln_bwd, pybind11::arg("dy"), pybind11::arg("s"),
        pybind11::arg("gamma"), pybind11::arg("mean"), pybind11::arg("rstd"),
        pybind11::arg("dgamma_out") = pybind11::none(),
        pybind11::arg("dbeta_out") = pybind11::none(),
        pybind11::arg("mask") = pybind11::none(),
        pybind11::arg("p") = 0.0);
  m.def("attn_fwd", &attn_fwd, pybind11::arg("q"), pybind11::arg("k"),
        pybind11::arg("v"), pybind11::arg("kv_pad"), pybind11::arg("causal"),
        pybind11::arg("scale"), pybind11::arg("trv") = 1);
  m.def("attn_bwd", &attn_bwd, pybind11::arg("q"), pybind11::arg("k"),
        pybind11::arg("v"), pybind11::arg("o"), pybind11::arg("dout"),
        pybind11::arg("lse"), pybind11::arg("kv_pad"), pybind11::arg("causal"),
        pybind11::arg("scale"), pybind11::arg("mode") = 0);
  m.def("embed_pe_fwd", &embed_pe_fwd);
  m.def("embed_pe_bwd", &embed_pe_bwd, pybind11::arg("dy"),
        pybind11::arg("tokens"), pybind11::arg("vocab"),
        pybind11::arg("out") = pybind11::none());
  m.def("dropout_fwd", &dropout_fwd, pybind11::arg("x"), pybind11::arg("p"),
        pybind11::arg("seed"), pybind11::arg("seed_t") = pybind11::none());
  m.def("dropout_bwd", &dropout_bwd);
  m.def("ce_fwd", &ce_fwd);
  m.def("ce_bwd", &ce_bwd);
  m.def("ce_fused", &ce_fused,
        "loss + gradient in one kernel (the grad sweep re-reads L2-hot "
        "rows); seed applied later by ce_scale (no-op at 1.0)");
  m.def("ce_scale", &ce_scale);
  m.def("adam_fused", &adam_fused);
  m.def("adam_fused_dev", &adam_fused_dev,
        "graph-capturable Adam: device step tensor + in-kernel Noam lr");
  m.def("argmax_lastdim", &argmax_lastdim);
  m.def("accuracy", &accuracy);
}
