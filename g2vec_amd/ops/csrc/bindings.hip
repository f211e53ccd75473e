// Torch bindings for the g2vec_amd gfx950 kernels + native host utilities.
// One translation unit: kernels are included below.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <cstdio>
#include <optional>
#include <cstring>
#include <fstream>
#include <sstream>
#include <string>
#include <vector>

#include "g2vec_kernels.hip"

#define CHECK_DEV(x) TORCH_CHECK((x).is_cuda(), #x " must be on the GPU")
#define CHECK_CONT(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")
#define CHECK_I32(x) TORCH_CHECK((x).scalar_type() == at::kInt, #x " must be int32")
#define CHECK_F32(x) TORCH_CHECK((x).scalar_type() == at::kFloat, #x " must be float32")

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

inline int grid_for(long long work_items, int per_block) {
  long long blocks = (work_items + per_block - 1) / per_block;
  if (blocks > (1 << 20)) blocks = 1 << 20;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

#define LAUNCH_CHECK()                                                       \
  do {                                                                       \
    hipError_t err = hipGetLastError();                                      \
    TORCH_CHECK(err == hipSuccess, "HIP launch failed: ",                    \
                hipGetErrorString(err));                                     \
  } while (0)

// ------------------------------------------------------------------ walks
std::vector<torch::Tensor> random_walks(torch::Tensor row_ptr, torch::Tensor col_idx,
                                        torch::Tensor weights, torch::Tensor sources,
                                        int64_t num_repetition, int64_t len_path,
                                        int64_t seed) {
  CHECK_DEV(row_ptr); CHECK_CONT(row_ptr); CHECK_I32(row_ptr);
  CHECK_DEV(col_idx); CHECK_CONT(col_idx); CHECK_I32(col_idx);
  CHECK_DEV(weights); CHECK_CONT(weights); CHECK_F32(weights);
  CHECK_DEV(sources); CHECK_CONT(sources); CHECK_I32(sources);
  TORCH_CHECK(len_path >= 1 && len_path <= 512, "len_path out of range");
  const long long n_src = sources.numel();
  const long long n_walks = n_src * num_repetition;
  auto opts_i = torch::TensorOptions().dtype(at::kInt).device(row_ptr.device());
  auto opts_l = torch::TensorOptions().dtype(at::kLong).device(row_ptr.device());
  auto nodes = torch::empty({n_walks, len_path}, opts_i);
  auto lengths = torch::empty({n_walks}, opts_i);
  auto hashes = torch::empty({n_walks}, opts_l);
  if (n_walks == 0) return {nodes, lengths, hashes};
  int tsize = 256;                   // LDS hash set: >= 2x path length, pow2
  while (tsize < 2 * (int)len_path) tsize <<= 1;
  const int wpb = 4;                 // 256 threads = 4 waves
  const size_t lds = (size_t)wpb * (len_path + tsize) * sizeof(int);
  hipLaunchKernelGGL(walk_kernel, dim3(grid_for(n_walks, wpb)), dim3(256), lds,
                     cur_stream(), row_ptr.data_ptr<int>(),
                     col_idx.data_ptr<int>(), weights.data_ptr<float>(),
                     sources.data_ptr<int>(),
                     (int)n_src, n_walks, (int)num_repetition, (int)len_path,
                     tsize, (uint64_t)seed,
                     nodes.data_ptr<int>(), lengths.data_ptr<int>(),
                     (long long*)hashes.data_ptr<int64_t>());
  LAUNCH_CHECK();
  return {nodes, lengths, hashes};
}

// ------------------------------------------------------------ CBOW (fast)
std::vector<torch::Tensor> cbow_fwd_scalar(torch::Tensor s, torch::Tensor genes,
                                           torch::Tensor offs, torch::Tensor labels,
                                           double inv_b, bool want_grad) {
  CHECK_DEV(s); CHECK_CONT(s); CHECK_F32(s);
  CHECK_DEV(genes); CHECK_CONT(genes); CHECK_I32(genes);
  CHECK_DEV(offs); CHECK_CONT(offs); CHECK_I32(offs);
  CHECK_DEV(labels); CHECK_CONT(labels); CHECK_F32(labels);
  const long long P = labels.numel();
  auto opts = torch::TensorOptions().dtype(at::kFloat).device(s.device());
  auto loss = torch::empty({P}, opts);
  auto correct = torch::empty({P}, opts);
  auto dO = want_grad ? torch::empty({P}, opts) : torch::empty({0}, opts);
  if (P == 0) return {loss, correct, dO};
  int grid = grid_for(P, 16);        // 16 paths per block (16-lane sub-waves)
  if (grid > 2048) grid = 2048;      // dispatch ramp of 8k+ tiny blocks costs
                                     // more than the grid-stride work itself
  hipLaunchKernelGGL(cbow_fwd_scalar_kernel, dim3(grid), dim3(256),
                     0, cur_stream(), s.data_ptr<float>(), genes.data_ptr<int>(),
                     offs.data_ptr<int>(), labels.data_ptr<float>(), P,
                     (float)inv_b, loss.data_ptr<float>(),
                     correct.data_ptr<float>(),
                     want_grad ? dO.data_ptr<float>() : nullptr);
  LAUNCH_CHECK();
  return {loss, correct, dO};
}

void cbow_eval_counts_(torch::Tensor s, torch::Tensor genes, torch::Tensor offs,
                       torch::Tensor labels, int64_t p_split,
                       torch::Tensor counts,
                       std::optional<torch::Tensor> dO, double inv_b) {
  if (dO) {
    CHECK_DEV(*dO); CHECK_CONT(*dO); CHECK_F32(*dO);
    TORCH_CHECK(dO->numel() >= p_split,
                "dO must cover the train split (p < p_split)");
  }
  CHECK_DEV(s); CHECK_CONT(s); CHECK_F32(s);
  CHECK_DEV(genes); CHECK_CONT(genes); CHECK_I32(genes);
  CHECK_DEV(offs); CHECK_CONT(offs); CHECK_I32(offs);
  CHECK_DEV(labels); CHECK_CONT(labels); CHECK_F32(labels);
  CHECK_DEV(counts); CHECK_CONT(counts); CHECK_F32(counts);
  TORCH_CHECK(counts.numel() == 2, "counts must have 2 elements");
  const long long P = labels.numel();
  if (P == 0) return;
  // LDS-staged variant (G2VEC_EVAL_LDS=<bytes> opts in when the whole s
  // table fits that per-block LDS budget): stages s in LDS so gathers are
  // bank-parallel ds_reads instead of L1 line-divergent loads, grid
  // capped (G2VEC_EVAL_LDS_GRID, default 1024) so each block amortizes
  // its stage. Bitwise-identical math. Measured (profiles/README.md,
  // round 2): wins the isolated microbench ~11% at ex_* shape (27.2 vs
  // 30.2 us, tools/bench_eval.py) but is NEUTRAL-to-slightly-worse
  // inside the real epoch chain (905 -> 893-902M paths/s same-box), so
  // it is OFF by default — kept as the documented measured alternative.
  const long long G = s.numel();
  const char* lds_env = getenv("G2VEC_EVAL_LDS");
  const long long lds_cap = lds_env ? atoll(lds_env) : 0;
  if (G * 4 <= lds_cap && G * 4 <= 158 * 1024) {   // 160 KB LDS hard limit
    const char* ge = getenv("G2VEC_EVAL_LDS_GRID");
    int grid = grid_for(P, 16);
    int gcap = ge ? atoi(ge) : 1024;
    if (gcap < 1) gcap = 1;         // malformed env must not 0-block-launch
    if (grid > gcap) grid = gcap;
    auto partials = torch::empty({grid, 2},
        torch::TensorOptions().dtype(at::kFloat).device(s.device()));
    hipLaunchKernelGGL(cbow_eval_counts_lds_kernel, dim3(grid),
                       dim3(256), (size_t)(G * 4), cur_stream(),
                       s.data_ptr<float>(),
                       genes.data_ptr<int>(), offs.data_ptr<int>(),
                       labels.data_ptr<float>(), P, (long long)p_split,
                       partials.data_ptr<float>(),
                       dO ? dO->data_ptr<float>() : nullptr, (float)inv_b,
                       (int)G);
    hipLaunchKernelGGL(fold_partials_kernel, dim3(1), dim3(256), 0,
                       cur_stream(), partials.data_ptr<float>(), grid,
                       counts.data_ptr<float>());
    LAUNCH_CHECK();
    return;
  }
  int grid = grid_for(P, 16);       // 16 paths per block (16-lane sub-waves)
  const char* gge = getenv("G2VEC_EVAL_GRID");   // sweep knob; default
  int gcap2 = gge ? atoi(gge) : 2048;            // 8192 waves fill the chip
  if (gcap2 < 1) gcap2 = 1;
  if (grid > gcap2) grid = gcap2;
  auto partials = torch::empty({grid, 2},
      torch::TensorOptions().dtype(at::kFloat).device(s.device()));
  hipLaunchKernelGGL(cbow_eval_counts_kernel, dim3(grid),
                     dim3(256), 0, cur_stream(), s.data_ptr<float>(),
                     genes.data_ptr<int>(), offs.data_ptr<int>(),
                     labels.data_ptr<float>(), P, (long long)p_split,
                     partials.data_ptr<float>(),
                     dO ? dO->data_ptr<float>() : nullptr, (float)inv_b);
  hipLaunchKernelGGL(fold_partials_kernel, dim3(1), dim3(256), 0,
                     cur_stream(), partials.data_ptr<float>(), grid,
                     counts.data_ptr<float>());
  LAUNCH_CHECK();
}

void cbow_eval_scan_(torch::Tensor s, torch::Tensor genes,
                     torch::Tensor pathid, torch::Tensor offs,
                     torch::Tensor labels, int64_t p_split, int64_t cap,
                     torch::Tensor piece, torch::Tensor counts,
                     std::optional<torch::Tensor> dO, double inv_b) {
  // instance-parallel fused eval (see eval_scan_kernel): piece buffer is
  // persistent (hipGraph-stable) and sized P * cap
  CHECK_DEV(s); CHECK_CONT(s); CHECK_F32(s);
  CHECK_DEV(genes); CHECK_CONT(genes); CHECK_I32(genes);
  CHECK_DEV(pathid); CHECK_CONT(pathid); CHECK_I32(pathid);
  CHECK_DEV(offs); CHECK_CONT(offs); CHECK_I32(offs);
  CHECK_DEV(labels); CHECK_CONT(labels); CHECK_F32(labels);
  CHECK_DEV(piece); CHECK_CONT(piece); CHECK_F32(piece);
  CHECK_DEV(counts); CHECK_CONT(counts); CHECK_F32(counts);
  const long long P = labels.numel();
  const long long nnz = genes.numel();
  TORCH_CHECK(piece.numel() >= P * cap, "piece buffer too small");
  if (dO) {
    CHECK_DEV(*dO); CHECK_CONT(*dO); CHECK_F32(*dO);
    TORCH_CHECK(dO->numel() >= p_split, "dO must cover the train split");
  }
  if (P == 0) return;
  int grid_a = grid_for(nnz, 256);
  const long long G = s.numel();
  int g_lds = 0;
  size_t lds = 0;
  if (G * 4 <= 48 * 1024) {
    // LDS-staged s: persistent blocks amortize the one-time 30 KB stage
    g_lds = (int)G;
    lds = (size_t)G * sizeof(float);
    if (grid_a > 1536) grid_a = 1536;
  }
  hipLaunchKernelGGL(eval_scan_kernel, dim3(grid_a), dim3(256), lds,
                     cur_stream(), s.data_ptr<float>(),
                     genes.data_ptr<int>(), pathid.data_ptr<int>(),
                     offs.data_ptr<int>(), nnz, (int)cap,
                     piece.data_ptr<float>(), g_lds);
  int grid_b = grid_for(P, 256);
  if (grid_b > 2048) grid_b = 2048;
  auto partials = torch::empty({grid_b, 2},
      torch::TensorOptions().dtype(at::kFloat).device(s.device()));
  hipLaunchKernelGGL(eval_finish_kernel, dim3(grid_b), dim3(256), 0,
                     cur_stream(), piece.data_ptr<float>(),
                     offs.data_ptr<int>(), labels.data_ptr<float>(), P,
                     (long long)p_split, (int)cap, partials.data_ptr<float>(),
                     dO ? dO->data_ptr<float>() : nullptr, (float)inv_b);
  hipLaunchKernelGGL(fold_partials_kernel, dim3(1), dim3(256), 0,
                     cur_stream(), partials.data_ptr<float>(), grid_b,
                     counts.data_ptr<float>());
  LAUNCH_CHECK();
}

void scatter_dO_det_(torch::Tensor inst_path, torch::Tensor seg_start,
                     torch::Tensor seg_gene, torch::Tensor dO,
                     torch::Tensor c) {
  // Accumulating out-arg variant: one slab's gene segments are reduced and
  // ADDED into c (caller zeroes it once). The slab-blocked caller issues
  // one launch per ~3 MB dO slice so each XCD's L2 holds the slice being
  // gathered instead of thrashing across the whole dO table.
  CHECK_DEV(inst_path); CHECK_CONT(inst_path); CHECK_I32(inst_path);
  CHECK_DEV(seg_start); CHECK_CONT(seg_start); CHECK_I32(seg_start);
  CHECK_DEV(seg_gene); CHECK_CONT(seg_gene); CHECK_I32(seg_gene);
  CHECK_DEV(dO); CHECK_CONT(dO); CHECK_F32(dO);
  CHECK_DEV(c); CHECK_CONT(c); CHECK_F32(c);
  const long long n_seg = seg_gene.numel();
  if (n_seg == 0) return;
  hipLaunchKernelGGL(scatter_do_det_kernel, dim3(grid_for(n_seg, 4)), dim3(256),
                     0, cur_stream(), inst_path.data_ptr<int>(),
                     seg_start.data_ptr<int>(), seg_gene.data_ptr<int>(),
                     (int)n_seg, dO.data_ptr<float>(), c.data_ptr<float>());
  LAUNCH_CHECK();
}

torch::Tensor scatter_dO_det(torch::Tensor inst_path, torch::Tensor seg_start,
                             torch::Tensor seg_gene, torch::Tensor dO,
                             int64_t n_genes) {
  CHECK_DEV(inst_path); CHECK_CONT(inst_path); CHECK_I32(inst_path);
  CHECK_DEV(seg_start); CHECK_CONT(seg_start); CHECK_I32(seg_start);
  CHECK_DEV(seg_gene); CHECK_CONT(seg_gene); CHECK_I32(seg_gene);
  CHECK_DEV(dO); CHECK_CONT(dO); CHECK_F32(dO);
  auto c = torch::zeros({n_genes},
                        torch::TensorOptions().dtype(at::kFloat).device(dO.device()));
  const long long n_seg = seg_gene.numel();
  if (n_seg == 0) return c;
  hipLaunchKernelGGL(scatter_do_det_kernel, dim3(grid_for(n_seg, 4)), dim3(256),
                     0, cur_stream(), inst_path.data_ptr<int>(),
                     seg_start.data_ptr<int>(), seg_gene.data_ptr<int>(),
                     (int)n_seg, dO.data_ptr<float>(), c.data_ptr<float>());
  LAUNCH_CHECK();
  return c;
}

void adam_rank1(torch::Tensor W, torch::Tensor m, torch::Tensor v,
                torch::Tensor c, torch::Tensor who, torch::Tensor lrt,
                double b1, double b2, double eps,
                std::optional<torch::Tensor> mO,
                std::optional<torch::Tensor> vO) {
  // mO/vO given: FUSED epoch tail — the same pass that streams the
  // pre-update W rows for the rank-1 Adam also emits per-block partials
  // of dW_ho = W_pre^T c, and one fold launch applies the who update
  // (replaces the separate gemv_cols + fold_cols + adam_dense chain and
  // its extra full W read; deterministic block order).
  CHECK_DEV(W); CHECK_CONT(W); CHECK_F32(W);
  CHECK_CONT(m); CHECK_CONT(v); CHECK_CONT(c); CHECK_CONT(who);
  CHECK_DEV(lrt); CHECK_F32(lrt);
  const long long G = W.size(0);
  const int h = (int)W.size(1);
  TORCH_CHECK(h % 4 == 0 && 256 % (h / 4) == 0 && h <= 1024,
              "hidden must be a multiple of 4 with h/4 dividing 256");
  const int rpb = 256 / (h / 4);   // rows per 256-thread block
  const bool fused = mO.has_value();
  int grid = grid_for(G, rpb);
  torch::Tensor partials;
  float* gw = nullptr;
  if (fused) {
    CHECK_DEV(*mO); CHECK_CONT(*mO); CHECK_F32(*mO);
    CHECK_DEV(*vO); CHECK_CONT(*vO); CHECK_F32(*vO);
    if (grid > 1024) grid = 1024;    // bound the partial table / fold cost
    partials = torch::empty({grid, h},
        torch::TensorOptions().dtype(at::kFloat).device(W.device()));
    gw = partials.data_ptr<float>();
  }
  hipLaunchKernelGGL(adam_rank1_kernel, dim3(grid), dim3(256), 0,
                     cur_stream(), W.data_ptr<float>(), m.data_ptr<float>(),
                     v.data_ptr<float>(), c.data_ptr<float>(),
                     who.data_ptr<float>(), G, h, lrt.data_ptr<float>(),
                     (float)b1, (float)b2, (float)eps, gw);
  if (fused) {
    hipLaunchKernelGGL(fold_gw_adam_kernel, dim3(grid_for(h, 4)), dim3(256),
                       0, cur_stream(), gw, grid, h, who.data_ptr<float>(),
                       mO->data_ptr<float>(), vO->data_ptr<float>(),
                       lrt.data_ptr<float>(), (float)b1, (float)b2,
                       (float)eps);
  }
  LAUNCH_CHECK();
}

void adam_dense(torch::Tensor W, torch::Tensor m, torch::Tensor v,
                torch::Tensor grad, torch::Tensor lrt, double b1, double b2,
                double eps) {
  CHECK_DEV(W); CHECK_CONT(W); CHECK_F32(W);
  CHECK_CONT(m); CHECK_CONT(v); CHECK_CONT(grad);
  CHECK_DEV(lrt); CHECK_F32(lrt);
  TORCH_CHECK(grad.numel() == W.numel(), "grad/W size mismatch");
  const long long n = W.numel();
  hipLaunchKernelGGL(adam_dense_kernel, dim3(grid_for(n, 256)), dim3(256), 0,
                     cur_stream(), W.data_ptr<float>(), m.data_ptr<float>(),
                     v.data_ptr<float>(), grad.data_ptr<float>(), n,
                     lrt.data_ptr<float>(), (float)b1, (float)b2, (float)eps);
  LAUNCH_CHECK();
}

// ---------------------------------------------------------- CBOW (general)
std::vector<torch::Tensor> cbow_fwd(torch::Tensor W, torch::Tensor who,
                                    torch::Tensor genes, torch::Tensor offs,
                                    torch::Tensor labels, double inv_b,
                                    bool want_grad, int64_t act) {
  CHECK_DEV(W); CHECK_CONT(W);
  CHECK_DEV(who); CHECK_CONT(who); CHECK_F32(who);
  CHECK_DEV(genes); CHECK_CONT(genes); CHECK_I32(genes);
  CHECK_DEV(offs); CHECK_CONT(offs); CHECK_I32(offs);
  CHECK_DEV(labels); CHECK_CONT(labels); CHECK_F32(labels);
  const bool bf16 = W.scalar_type() == at::kBFloat16;
  const bool fp16 = W.scalar_type() == at::kHalf;
  TORCH_CHECK(bf16 || fp16 || W.scalar_type() == at::kFloat,
              "W must be f32, bf16 or fp16");
  const long long P = labels.numel();
  const int h = (int)W.size(1);
  const int hpl = h / 64;
  TORCH_CHECK(h % 64 == 0 && hpl >= 1 && hpl <= 16 &&
              (hpl & (hpl - 1)) == 0, "hidden must be 64*{1,2,4,8,16}");
  auto opts = torch::TensorOptions().dtype(at::kFloat).device(W.device());
  auto loss = torch::empty({P}, opts);
  auto correct = torch::empty({P}, opts);
  auto dO = want_grad ? torch::empty({P}, opts) : torch::empty({0}, opts);
  auto H = want_grad ? torch::empty({P, h}, opts) : torch::empty({0}, opts);
  if (P == 0) return {loss, correct, dO, H};
  const int grid = grid_for(P, 4);
  float* Hp = want_grad ? H.data_ptr<float>() : nullptr;
  float* dOp = want_grad ? dO.data_ptr<float>() : nullptr;

#define FWD_CASE(WT, HPL, PTR)                                                \
  hipLaunchKernelGGL((cbow_fwd_kernel<WT, HPL>), dim3(grid), dim3(256), 0,    \
                     cur_stream(), PTR, who.data_ptr<float>(),                \
                     genes.data_ptr<int>(), offs.data_ptr<int>(),             \
                     labels.data_ptr<float>(), P, (float)inv_b, h, Hp,        \
                     loss.data_ptr<float>(), correct.data_ptr<float>(), dOp,  \
                     (int)act)
  if (bf16) {
    const bf16_bits* Wp = (const bf16_bits*)W.data_ptr<at::BFloat16>();
    switch (hpl) {
      case 1: FWD_CASE(bf16_bits, 1, Wp); break;
      case 2: FWD_CASE(bf16_bits, 2, Wp); break;
      case 4: FWD_CASE(bf16_bits, 4, Wp); break;
      case 8: FWD_CASE(bf16_bits, 8, Wp); break;
      default: FWD_CASE(bf16_bits, 16, Wp); break;
    }
  } else if (fp16) {
    const fp16_bits* Wp = (const fp16_bits*)W.data_ptr<at::Half>();
    switch (hpl) {
      case 1: FWD_CASE(fp16_bits, 1, Wp); break;
      case 2: FWD_CASE(fp16_bits, 2, Wp); break;
      case 4: FWD_CASE(fp16_bits, 4, Wp); break;
      case 8: FWD_CASE(fp16_bits, 8, Wp); break;
      default: FWD_CASE(fp16_bits, 16, Wp); break;
    }
  } else {
    const float* Wp = W.data_ptr<float>();
    switch (hpl) {
      case 1: FWD_CASE(float, 1, Wp); break;
      case 2: FWD_CASE(float, 2, Wp); break;
      case 4: FWD_CASE(float, 4, Wp); break;
      case 8: FWD_CASE(float, 8, Wp); break;
      default: FWD_CASE(float, 16, Wp); break;
    }
  }
#undef FWD_CASE
  LAUNCH_CHECK();
  return {loss, correct, dO, H};
}

torch::Tensor cbow_bwd_rows(torch::Tensor who, torch::Tensor inst_path,
                            torch::Tensor seg_start, torch::Tensor seg_gene,
                            torch::Tensor dO, int64_t n_genes,
                            std::optional<torch::Tensor> Hpre) {
  // Hpre: pre-activation H for the ReLU backward mask (absent = linear)
  CHECK_DEV(who); CHECK_CONT(who); CHECK_F32(who);
  CHECK_DEV(inst_path); CHECK_CONT(inst_path); CHECK_I32(inst_path);
  CHECK_DEV(seg_start); CHECK_CONT(seg_start); CHECK_I32(seg_start);
  CHECK_DEV(seg_gene); CHECK_CONT(seg_gene); CHECK_I32(seg_gene);
  CHECK_DEV(dO); CHECK_CONT(dO); CHECK_F32(dO);
  if (Hpre) { CHECK_DEV(*Hpre); CHECK_CONT(*Hpre); CHECK_F32(*Hpre); }
  const int h = (int)who.numel();
  const int hpl = h / 64;
  TORCH_CHECK(h % 64 == 0 && hpl >= 1 && hpl <= 16 &&
              (hpl & (hpl - 1)) == 0, "hidden must be 64*{1,2,4,8,16}");
  auto dW = torch::zeros({n_genes, h},
                         torch::TensorOptions().dtype(at::kFloat).device(dO.device()));
  const int n_seg = (int)seg_gene.numel();
  if (n_seg == 0) return dW;
  const int grid = grid_for(n_seg, 4);
#define BWD_CASE(HPL)                                                         \
  hipLaunchKernelGGL((cbow_bwd_rows_det_kernel<HPL>), dim3(grid), dim3(256),  \
                     0, cur_stream(), who.data_ptr<float>(),                  \
                     inst_path.data_ptr<int>(), seg_start.data_ptr<int>(),    \
                     seg_gene.data_ptr<int>(), n_seg,                         \
                     dO.data_ptr<float>(), h, dW.data_ptr<float>(),           \
                     Hpre ? Hpre->data_ptr<float>() : nullptr)
  switch (hpl) {
    case 1: BWD_CASE(1); break;
    case 2: BWD_CASE(2); break;
    case 4: BWD_CASE(4); break;
    case 8: BWD_CASE(8); break;
    default: BWD_CASE(16); break;
  }
#undef BWD_CASE
  LAUNCH_CHECK();
  return dW;
}

// ------------------------------------------------------------------- PCC
torch::Tensor pcc_edges(torch::Tensor zt, torch::Tensor edge_idx,
                        int64_t n_group) {
  CHECK_DEV(zt); CHECK_CONT(zt); CHECK_F32(zt);
  CHECK_DEV(edge_idx); CHECK_CONT(edge_idx); CHECK_I32(edge_idx);
  const long long E = edge_idx.size(0);
  const int S = (int)zt.size(1);
  auto out = torch::empty({E},
                          torch::TensorOptions().dtype(at::kFloat).device(zt.device()));
  if (E == 0) return out;
  hipLaunchKernelGGL(pcc_edges_kernel, dim3(grid_for(E, 4)), dim3(256), 0,
                     cur_stream(), zt.data_ptr<float>(),
                     edge_idx.data_ptr<int>(), E, S, 1.0f / (float)n_group,
                     out.data_ptr<float>());
  LAUNCH_CHECK();
  return out;
}

torch::Tensor corr_gemm(torch::Tensor zt, int64_t n_group) {
  CHECK_DEV(zt); CHECK_CONT(zt); CHECK_F32(zt);
  // the kernel body is compiled only for gfx950 (MFMA builtins): on any
  // other arch the launch would be a no-op returning uninitialized C —
  // trap here instead of silently emitting garbage PCC weights
  {
    hipDeviceProp_t prop;
    int dev = 0;
    (void)hipGetDevice(&dev);
    (void)hipGetDeviceProperties(&prop, dev);
    TORCH_CHECK(std::string(prop.gcnArchName).rfind("gfx950", 0) == 0,
                "corr_gemm: MFMA kernel is gfx950-only, device reports ",
                prop.gcnArchName, " (use pcc_mode=edge)");
  }
  const int G = (int)zt.size(0);
  const int S = (int)zt.size(1);
  const int S4 = ((S + 3) / 4) * 4;
  const size_t lds = 2ull * 64 * S4 * sizeof(float);
  TORCH_CHECK(lds <= 160 * 1024,
              "corr_gemm: S too large for the LDS-staged tile (use pcc_mode=edge)");
  auto C = torch::empty({G, G},
                        torch::TensorOptions().dtype(at::kFloat).device(zt.device()));
  const int ntile = (G + 63) / 64;
  hipLaunchKernelGGL(corr_gemm_kernel, dim3(ntile * ntile), dim3(256), lds,
                     cur_stream(), zt.data_ptr<float>(), C.data_ptr<float>(),
                     G, S, S4, 1.0f / (float)n_group);
  LAUNCH_CHECK();
  return C;
}

void trunc_normal_(torch::Tensor out, double std, int64_t seed) {
  CHECK_DEV(out); CHECK_CONT(out); CHECK_F32(out);
  const long long n = (long long)out.numel();
  if (n == 0) return;
  hipLaunchKernelGGL(trunc_normal_kernel, dim3(grid_for(n, 256)), dim3(256),
                     0, cur_stream(), out.data_ptr<float>(), n, (float)std,
                     (uint64_t)seed);
  LAUNCH_CHECK();
}

void gemv_rows_(torch::Tensor W, torch::Tensor x, torch::Tensor out) {
  CHECK_DEV(W); CHECK_CONT(W); CHECK_F32(W);
  CHECK_DEV(x); CHECK_CONT(x); CHECK_F32(x);
  CHECK_DEV(out); CHECK_CONT(out); CHECK_F32(out);
  const long long G = W.size(0);
  const int h = (int)W.size(1);
  const int hpl = h / 64;
  TORCH_CHECK(h % 64 == 0 && hpl >= 1 && hpl <= 16 && (hpl & (hpl - 1)) == 0,
              "hidden must be 64*{1,2,4,8,16}");
  TORCH_CHECK(out.numel() == G && x.numel() == h, "gemv_rows shape mismatch");
  int grid = grid_for(G, 4);
  if (grid > 8192) grid = 8192;
#define GEMVR_CASE(HPL)                                                       \
  hipLaunchKernelGGL((gemv_rows_kernel<HPL>), dim3(grid), dim3(256), 0,       \
                     cur_stream(), W.data_ptr<float>(), x.data_ptr<float>(),  \
                     G, h, out.data_ptr<float>())
  switch (hpl) {
    case 1: GEMVR_CASE(1); break;
    case 2: GEMVR_CASE(2); break;
    case 4: GEMVR_CASE(4); break;
    case 8: GEMVR_CASE(8); break;
    default: GEMVR_CASE(16); break;
  }
#undef GEMVR_CASE
  LAUNCH_CHECK();
}

void gemv_cols_(torch::Tensor W, torch::Tensor c, torch::Tensor out) {
  CHECK_DEV(W); CHECK_CONT(W); CHECK_F32(W);
  CHECK_DEV(c); CHECK_CONT(c); CHECK_F32(c);
  CHECK_DEV(out); CHECK_CONT(out); CHECK_F32(out);
  const long long G = W.size(0);
  const int h = (int)W.size(1);
  const int cpt = (h >= 256) ? h / 256 : 1;
  TORCH_CHECK(h % 64 == 0 && cpt >= 1 && cpt <= 4 && (cpt & (cpt - 1)) == 0,
              "gemv_cols needs hidden 64*k with h/256 in {<=1,2,4}");
  TORCH_CHECK(out.numel() == h && c.numel() == G, "gemv_cols shape mismatch");
  // ~8 rows per block: small G must still spread over the chip (8 blocks
  // for G=7.5k measured 78 us of serial row-walking; 941 blocks ~10 us)
  int grid = (int)((G + 7) / 8);
  if (grid > 2048) grid = 2048;
  if (grid < 1) grid = 1;
  auto partials = torch::empty({grid, h},
      torch::TensorOptions().dtype(at::kFloat).device(W.device()));
#define GEMVC_CASE(CPT)                                                       \
  hipLaunchKernelGGL((gemv_cols_kernel<CPT>), dim3(grid), dim3(256), 0,       \
                     cur_stream(), W.data_ptr<float>(), c.data_ptr<float>(),  \
                     G, h, partials.data_ptr<float>())
  switch (cpt) {
    case 1: GEMVC_CASE(1); break;
    case 2: GEMVC_CASE(2); break;
    default: GEMVC_CASE(4); break;
  }
#undef GEMVC_CASE
  hipLaunchKernelGGL(fold_cols_kernel, dim3((h + 3) / 4), dim3(256), 0,
                     cur_stream(), partials.data_ptr<float>(), grid, h,
                     out.data_ptr<float>());
  LAUNCH_CHECK();
}

torch::Tensor bf16_copy(torch::Tensor src) {
  CHECK_DEV(src); CHECK_CONT(src); CHECK_F32(src);
  auto out = torch::empty_like(src, src.options().dtype(at::kBFloat16));
  const long long n = src.numel();
  if (n)
    hipLaunchKernelGGL(f32_to_bf16_kernel, dim3(grid_for(n, 256)), dim3(256),
                       0, cur_stream(), src.data_ptr<float>(),
                       (uint16_t*)out.data_ptr<at::BFloat16>(), n);
  LAUNCH_CHECK();
  return out;
}

// ------------------------------------------------- native host TSV parser
std::tuple<std::vector<std::string>, std::vector<std::string>, torch::Tensor>
parse_expression_tsv(const std::string& path) {
  std::ifstream f(path);
  TORCH_CHECK(f.good(), "cannot open ", path);
  std::string line;
  TORCH_CHECK(std::getline(f, line), "empty file ", path);
  std::vector<std::string> samples;
  {
    size_t pos = line.find('\t');
    TORCH_CHECK(pos != std::string::npos, "bad header in ", path);
    size_t start = pos + 1;
    while (start <= line.size()) {
      size_t next = line.find('\t', start);
      if (next == std::string::npos) {
        std::string v = line.substr(start);
        while (!v.empty() && (v.back() == '\r' || v.back() == '\n')) v.pop_back();
        if (!v.empty()) samples.push_back(v);
        break;
      }
      samples.push_back(line.substr(start, next - start));
      start = next + 1;
    }
  }
  const size_t S = samples.size();
  std::vector<std::string> genes;
  std::vector<float> vals;
  vals.reserve(S * 4096);
  long long ln = 1;
  while (std::getline(f, line)) {
    ++ln;
    if (line.empty()) continue;
    const char* p = line.c_str();
    const char* tab = std::strchr(p, '\t');
    if (!tab) continue;               // lenient: short rows skipped (like
                                      // the Python fallback)
    genes.emplace_back(p, tab - p);
    const char* q = tab + 1;
    for (size_t i = 0; i < S; ++i) {
      TORCH_CHECK(*q != '\0',
                  path, ":", ln, ": gene '", genes.back(), "' has ", i,
                  " values, expected ", S, " (one per sample column)");
      char* end = nullptr;
      const float v = std::strtof(q, &end);
      TORCH_CHECK(end != q && (*end == '\t' || *end == '\0' ||
                               *end == '\r' || *end == '\n'),
                  path, ":", ln, ": non-numeric expression value for gene '",
                  genes.back(), "'");
      vals.push_back(v);
      const bool last = (i + 1 == S);
      TORCH_CHECK(last || *end == '\t',
                  path, ":", ln, ": gene '", genes.back(), "' has ", i + 1,
                  " values, expected ", S, " (one per sample column)");
      q = (*end == '\t') ? end + 1 : end;
    }
    // trailing extra columns = ragged row
    while (*q == '\r' || *q == '\n') ++q;
    TORCH_CHECK(*q == '\0',
                path, ":", ln, ": gene '", genes.back(),
                "' has more than ", S, " values (one per sample column)");
  }
  TORCH_CHECK(!genes.empty(), path,
              ": no expression rows (empty or header-only file)");
  const size_t G = genes.size();
  auto t = torch::from_blob(vals.data(), {(long long)G, (long long)S},
                            torch::TensorOptions().dtype(at::kFloat))
               .clone();
  // gene-wise -> sample-wise (like the reference transpose, G2Vec.py:498)
  return {genes, samples, t.t().contiguous()};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("random_walks", &random_walks, "CSR biased random walks (gfx950)");
  m.def("cbow_fwd_scalar", &cbow_fwd_scalar, "scalar CBOW forward + loss");
  m.def("scatter_dO_det", &scatter_dO_det, "deterministic c = X^T dO");
  m.def("scatter_dO_det_", &scatter_dO_det_,
        "deterministic c += X^T dO (one slab, out-arg)");
  m.def("cbow_eval_scan_", &cbow_eval_scan_,
        py::arg("s"), py::arg("genes"), py::arg("pathid"), py::arg("offs"),
        py::arg("labels"), py::arg("p_split"), py::arg("cap"),
        py::arg("piece"), py::arg("counts"),
        py::arg("dO") = py::none(), py::arg("inv_b") = 1.0,
        "instance-parallel fused eval (segmented scan + finish)");
  m.def("cbow_eval_counts_", &cbow_eval_counts_,
        py::arg("s"), py::arg("genes"), py::arg("offs"), py::arg("labels"),
        py::arg("p_split"), py::arg("counts"),
        py::arg("dO") = py::none(), py::arg("inv_b") = 1.0,
        "fused train/val correct-count eval (in-place counts[2])");
  m.def("adam_rank1", &adam_rank1,
        py::arg("W"), py::arg("m"), py::arg("v"), py::arg("c"),
        py::arg("who"), py::arg("lrt"), py::arg("b1"), py::arg("b2"),
        py::arg("eps"), py::arg("mO") = py::none(), py::arg("vO") = py::none(),
        "TF1 Adam, rank-1 grad (mO/vO given: fused dW_ho + who update)");
  m.def("adam_dense", &adam_dense, "TF1 Adam, dense grad");
  m.def("cbow_fwd", &cbow_fwd,
        py::arg("W"), py::arg("who"), py::arg("genes"), py::arg("offs"),
        py::arg("labels"), py::arg("inv_b"), py::arg("want_grad"),
        py::arg("act") = 0,
        "row-gather CBOW forward (act: 0 linear, 1 ReLU)");
  m.def("cbow_bwd_rows", &cbow_bwd_rows,
        py::arg("who"), py::arg("inst_path"), py::arg("seg_start"),
        py::arg("seg_gene"), py::arg("dO"), py::arg("n_genes"),
        py::arg("Hpre") = py::none(),
        "deterministic scatter CBOW backward (Hpre = ReLU mask source)");
  m.def("pcc_edges", &pcc_edges, "per-edge |PCC|");
  m.def("corr_gemm", &corr_gemm, "MFMA f32 correlation GEMM");
  m.def("trunc_normal_", &trunc_normal_,
        "seeded +-2sigma truncated-normal fill (K9)");
  m.def("gemv_rows_", &gemv_rows_, "s = W @ x (wave-per-row GEMV, out-arg)");
  m.def("gemv_cols_", &gemv_cols_, "out = W^T c (partials + fold, out-arg)");
  m.def("bf16_copy", &bf16_copy, "f32 -> bf16 cast kernel");
  m.def("parse_expression_tsv", &parse_expression_tsv,
        "native expression TSV parser");
}
