// Python bindings for the NornicDB-AMD native kernel library (_C).
#include <torch/extension.h>

// vector_ops.hip
void l2_normalize_(at::Tensor x);
void fill_random_unit_(at::Tensor x, long long row_base, long long seed);
std::tuple<at::Tensor, at::Tensor> knn_gemv(at::Tensor db, at::Tensor q,
                                            long long row_base, int k_out);

// knn_mfma.hip
long long knn_bm_value();
#define KNN_BM_VALUE knn_bm_value()
std::tuple<at::Tensor, at::Tensor> knn_mfma(at::Tensor db, at::Tensor q,
                                            long long row_base, int k_out);

// gemm.hip
at::Tensor gemm_nt(at::Tensor a, at::Tensor w, c10::optional<at::Tensor> bias,
                   long long act);

// kmeans.hip
std::tuple<at::Tensor, at::Tensor> kmeans_assign(at::Tensor x, at::Tensor cent,
                                                 at::Tensor cnorm2);
std::tuple<at::Tensor, at::Tensor> kmeans_accum(at::Tensor x, at::Tensor assign,
                                                long long k);
std::tuple<at::Tensor, at::Tensor> kmeans_finalize(at::Tensor sums,
                                                   at::Tensor counts,
                                                   at::Tensor old_c);
void kmeanspp_update(at::Tensor x, at::Tensor c, double cn2, at::Tensor d2);
void kmeans_point_update(at::Tensor cent, at::Tensor counts, at::Tensor xv,
                         long long c, long long sign);

// graph.hip
at::Tensor pagerank_contrib(at::Tensor rank, at::Tensor outdeg);
at::Tensor pagerank_gather(at::Tensor row_ptr, at::Tensor col_idx,
                           at::Tensor contrib, double damping, double base);
void bfs_level(at::Tensor row_ptr, at::Tensor col_idx, at::Tensor dist,
               at::Tensor changed, long long row_base, long long level);
at::Tensor labelprop_step(at::Tensor row_ptr, at::Tensor col_idx,
                          at::Tensor labels, at::Tensor changed,
                          long long row_base);
void wcc_hook(at::Tensor row_ptr, at::Tensor col_idx, at::Tensor comp,
              at::Tensor changed, long long row_base);

// packstream.cpp (CPU)
pybind11::bytes ps_pack(pybind11::object v);
pybind11::object ps_unpack(pybind11::buffer data,
                           pybind11::object structure_factory);

// encoder_ops.hip
at::Tensor add_layernorm(at::Tensor a, c10::optional<at::Tensor> b,
                         at::Tensor gamma, at::Tensor beta, double eps);
at::Tensor bias_gelu(at::Tensor x, at::Tensor bias);
at::Tensor mean_pool_l2norm(at::Tensor x, c10::optional<at::Tensor> mask);
at::Tensor flash_attn_nc(at::Tensor q, at::Tensor k, at::Tensor v);
void decode_step(at::Tensor layer_ptrs, at::Tensor x, at::Tensor q,
                 at::Tensor attn, at::Tensor h, at::Tensor rope_cos,
                 at::Tensor rope_sin, long long n_layers, long long hidden,
                 long long n_heads, long long n_kv, long long hd,
                 long long inter, long long max_len, double rms_eps,
                 long long pos);
double sync_bench(long long iters, long long which, long long grid,
                  at::Tensor scratch);
// knn_fp8.hip
std::tuple<at::Tensor, at::Tensor> knn_fp8(at::Tensor db, at::Tensor q,
                                           long long row_base, int k_out);
std::tuple<at::Tensor, at::Tensor> knn_i8(at::Tensor db, at::Tensor sa,
                                          at::Tensor q, at::Tensor sq,
                                          long long row_base, int k_out);
void decode_tokens(at::Tensor layer_ptrs, at::Tensor x, at::Tensor q,
                   at::Tensor attn, at::Tensor h, at::Tensor rope_cos,
                   at::Tensor rope_sin, at::Tensor embed_w,
                   at::Tensor norm_w, at::Tensor lm_w, at::Tensor out,
                   at::Tensor n_done, at::Tensor pmax, at::Tensor pidx,
                   long long n_layers, long long hidden, long long n_heads,
                   long long n_kv, long long hd, long long inter,
                   long long max_len, double rms_eps, long long pos,
                   long long start_tok, long long n_toks, long long eos);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "NornicDB-AMD CDNA4 (gfx950) native kernels";
  m.def("l2_normalize_", &l2_normalize_, "In-place row L2 normalize (bf16/f32)");
  m.def("fill_random_unit_", &fill_random_unit_,
        "Fill with deterministic unit-norm pseudo-gaussian rows (bf16)",
        py::arg("x"), py::arg("row_base") = 0, py::arg("seed") = 0x6e6f726eLL);
  m.def("knn_gemv", &knn_gemv,
        "Fused cosine score + top-k for <=16 queries (bf16 db)",
        py::arg("db"), py::arg("q"), py::arg("row_base") = 0,
        py::arg("k_out") = 10);
  m.attr("KNN_BM") = KNN_BM_VALUE;
  m.def("knn_mfma", &knn_mfma,
        "Fused MFMA cosine score + top-k, 256-query batches (bf16 db)",
        py::arg("db"), py::arg("q"), py::arg("row_base") = 0,
        py::arg("k_out") = 10);
  m.def("gemm_nt", &gemm_nt,
        "C = A[M,K] @ W[N,K]^T + bias (+GELU), bf16 MFMA 256x256 8-phase",
        py::arg("a"), py::arg("w"), py::arg("bias") = c10::nullopt,
        py::arg("act") = 0);
  m.def("add_layernorm", &add_layernorm, "LN(a+b)*gamma+beta fused (bf16)",
        py::arg("a"), py::arg("b"), py::arg("gamma"), py::arg("beta"),
        py::arg("eps") = 1e-5);
  m.def("bias_gelu", &bias_gelu, "gelu(x+bias) fused (bf16)");
  m.def("mean_pool_l2norm", &mean_pool_l2norm,
        "masked mean-pool + L2 norm (bf16 -> fp32)",
        py::arg("x"), py::arg("mask") = c10::nullopt);
  m.def("flash_attn_nc", &flash_attn_nc,
        "non-causal flash attention fwd, head_dim 64 (bf16)");
  m.def("decode_step", &decode_step,
        "fused cooperative single-token decode (all layers, one launch)",
        py::arg("layer_ptrs"), py::arg("x"), py::arg("q"), py::arg("attn"),
        py::arg("h"), py::arg("rope_cos"), py::arg("rope_sin"),
        py::arg("n_layers"), py::arg("hidden"), py::arg("n_heads"),
        py::arg("n_kv"), py::arg("hd"), py::arg("inter"),
        py::arg("max_len"), py::arg("rms_eps"), py::arg("pos"));
  m.def("decode_tokens", &decode_tokens,
        "fused cooperative multi-token greedy decode (embed + layers + "
        "lm_head argmax, whole generation loop in one launch)",
        py::arg("layer_ptrs"), py::arg("x"), py::arg("q"), py::arg("attn"),
        py::arg("h"), py::arg("rope_cos"), py::arg("rope_sin"),
        py::arg("embed_w"), py::arg("norm_w"), py::arg("lm_w"),
        py::arg("out"), py::arg("n_done"), py::arg("pmax"), py::arg("pidx"),
        py::arg("n_layers"), py::arg("hidden"), py::arg("n_heads"),
        py::arg("n_kv"), py::arg("hd"), py::arg("inter"),
        py::arg("max_len"), py::arg("rms_eps"), py::arg("pos"),
        py::arg("start_tok"), py::arg("n_toks"), py::arg("eos") = -1);
  m.def("knn_i8", &knn_i8,
        "Fused i8 MFMA score + top-k over a symmetric int8 corpus with "
        "per-row scales (score = i32 dot * sa[row] * sq[col])",
        py::arg("db"), py::arg("sa"), py::arg("q"), py::arg("sq"),
        py::arg("row_base") = 0, py::arg("k_out") = 10);
  m.def("knn_fp8", &knn_fp8,
        "Fused MFMA cosine score + top-k over an FP8 e4m3fn corpus "
        "(uint8 carrier), 256-query batches",
        py::arg("db"), py::arg("q"), py::arg("row_base") = 0,
        py::arg("k_out") = 10);
  m.def("sync_bench", &sync_bench,
        "grid-barrier microbenchmark: ms for `iters` barriers "
        "(which=0 cg::grid.sync, 1 two-level custom)",
        py::arg("iters"), py::arg("which"), py::arg("grid"),
        py::arg("scratch"));
  m.def("kmeans_assign", &kmeans_assign,
        "fused distance+argmin assignment (bf16 x/centroids)");
  m.def("kmeans_accum", &kmeans_accum, "atomic centroid accumulate");
  m.def("kmeans_finalize", &kmeans_finalize,
        "sums/counts -> centroids + drift^2 (empty keep old)");
  m.def("kmeanspp_update", &kmeanspp_update,
        "d2 = min(d2, dist2(x, new_seed))");
  m.def("kmeans_point_update", &kmeans_point_update,
        "incremental single-point centroid update (+1 add / -1 remove)");
  m.def("pagerank_contrib", &pagerank_contrib, "rank/outdeg elementwise");
  m.def("pagerank_gather", &pagerank_gather,
        "CSR pull-gather pagerank iteration (wave per row)");
  m.def("bfs_level", &bfs_level, "BFS frontier level expansion");
  m.def("labelprop_step", &labelprop_step, "label propagation step");
  m.def("wcc_hook", &wcc_hook, "connected-components hook step");
  m.def("ps_pack", &ps_pack, "PackStream encode (native)");
  m.def("ps_unpack", &ps_unpack, "PackStream decode (native)",
        py::arg("data"), py::arg("structure_factory"));
}
