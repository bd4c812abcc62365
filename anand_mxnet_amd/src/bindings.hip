// Python bindings for the MI355X-native kernel library (_C extension).
// Plays the role of the reference's flat C ABI (src/c_api/) — here the
// native boundary is a torch extension: tensors in, stream-ordered HIP
// kernel launches out, zero Python in the hot loop.
#include <torch/extension.h>
#include <pybind11/numpy.h>

#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>

#include <unordered_map>
#include <vector>

#include "kernels/common.h"  // bf16 typedef, ZMap batch map

namespace mxa {
// gemm_tn.hip
void gemm_tn_bf16(const bf16*, const bf16*, bf16*, const float*, int, int,
                  int, long, long, long, long, long, long, int, bool,
                  const bf16*, hipStream_t, float* stats = nullptr);
void gemm_tn_splitk_bf16(const bf16*, const bf16*, bf16*, float*,
                         const float*, int, int, int, bool, const bf16*,
                         hipStream_t);
bool conv2d_fwd_bf16(const bf16*, const bf16*, bf16*, float*, const float*,
                     int, int, int, int, int, int, int, int, int, int, int,
                     int, int, int, int, bool, const bf16*, hipStream_t,
                     float* stats = nullptr);
void conv2d_dgrad_bf16(const bf16*, const bf16*, bf16*, float*, int, int,
                       int, int, int, int, int, int, int, int, int, int,
                       int, int, int, const bf16*, hipStream_t);
// gemm_tn8.hip
void gemm_tn8_bf16(const bf16*, const bf16*, bf16*, const float*, int, int,
                   int, long, long, long, long, long, long, int, bool,
                   const bf16*, hipStream_t);
void gemm_tn8h_bf16(const bf16*, const bf16*, bf16*, const float*, int,
                    int, int, long, long, long, long, long, long, int,
                    bool, const bf16*, hipStream_t);
// gemm_wg.hip (slot-buffer split-K + fold epilogue)
int gemm_wg_nslots(long, int, int, bool);
void gemm_wg_bf16(const bf16*, const bf16*, float*, bf16*, bool, long, int,
                  int, long, long, const bf16*, hipStream_t,
                  float* biasg = nullptr);
int conv2d_wgrad_nslots(long, int, int, int, int);
void conv2d_wgrad_bf16(const bf16*, const bf16*, float*, bf16*, int, int,
                       int, int, int, int, int, int, int, int, int, int,
                       int, int, int, const bf16*, hipStream_t);
bool conv_wg_c9_eligible(int, int, int, int, int, int, int, int, int, int,
                         int, int, int, int);
long conv_wg_c9_xp_elems(int, int, int, int);
void conv_wg_pad_image(const bf16*, bf16*, int, int, int, int, hipStream_t);
int conv2d_wgrad_c9_nslots(int, int, int, int, int);
void conv2d_wgrad_c9_bf16(const bf16*, const bf16*, float*, bf16*, int, int,
                          int, int, int, const bf16*, hipStream_t);
// bn_pool.hip
void bn_fwd_train(const bf16*, bf16*, float*, float*, float*, float*, float*,
                  float*, const float*, const float*, long, int, float,
                  float, bool, const bf16*, hipStream_t,
                  const float* ext_sums = nullptr);
void bn_fwd_infer(const bf16*, bf16*, const float*, const float*,
                  const float*, const float*, long, int, float, bool,
                  const bf16*, hipStream_t);
void bn_bwd(const bf16*, const bf16*, bf16*, const float*, const float*,
            const float*, const float*, float*, float*, long, int, bool,
            hipStream_t);
int bn_gy(long, int);
void maxpool_nhwc(const bf16*, bf16*, unsigned char*, int, int, int, int,
                  int, int, int, int, int, int, int, int, hipStream_t);
void maxpool_nhwc_bwd(const bf16*, const unsigned char*, bf16*, int, int,
                      int, int, int, int, int, int, int, int, int, int,
                      hipStream_t);
void avgpool_nhwc(const bf16*, bf16*, int, int, int, int, int, int, int,
                  int, int, int, int, int, bool, hipStream_t);
void gap_nhwc(const bf16*, bf16*, int, long, int, hipStream_t);
void gap_nhwc_bwd(const bf16*, bf16*, int, long, int, hipStream_t);
// softmax_norm.hip
void softmax_fwd_bf16(const bf16*, bf16*, long, int, bool, float,
                      hipStream_t);
void softmax_masked_fwd_bf16(const bf16*, bf16*, long, int, const int*,
                             float,
                             hipStream_t);
void softmax_bwd_bf16(const bf16*, const bf16*, bf16*, long, int, bool,
                      float, hipStream_t);
void softmax_dropout_fwd_bf16(const bf16*, bf16*, bf16*, unsigned char*,
                              long, int, const int*, float,
                              const unsigned long long*,
                              unsigned long long, float, hipStream_t);
void softmax_dropout_bwd_bf16(const bf16*, const bf16*,
                              const unsigned char*, bf16*, long, int,
                              float, float, hipStream_t);
void layernorm_fwd_bf16(const bf16*, bf16*, const float*, const float*,
                        float*, float*, long, int, float, hipStream_t);
void layernorm_bwd_bf16(const bf16*, const bf16*, bf16*, const float*,
                        const float*, const float*, float*, float*, long,
                        int, hipStream_t);
// misc.hip
void dropout_fwd_bf16(const bf16*, bf16*, unsigned char*, long,
                      const unsigned long long*, unsigned long long, float,
                      hipStream_t);
void rng_bump_launch(unsigned long long*, hipStream_t);
void dropout_bwd_bf16(const bf16*, const unsigned char*, bf16*, long, float,
                      hipStream_t);
void embedding_fwd_bf16(const long*, const bf16*, bf16*, long, int, long,
                        hipStream_t);
void embedding_bwd_bf16(const long*, const bf16*, float*, long, int, long,
                        hipStream_t);
void colsum_bf16(const bf16*, float*, long, int, hipStream_t);
void im2col_stem_bf16(const bf16*, bf16*, int, int, int, int, int, int,
                      int, int, int, int, int, int, int, int, int, int,
                      hipStream_t);
void dwconv_fwd_bf16(const bf16*, const bf16*, const float*, bf16*, int,
                     int, int, int, int, int, int, int, int, int, int,
                     int, int, int, hipStream_t);
void dwconv_dgrad_bf16(const bf16*, const bf16*, bf16*, int, int, int,
                       int, int, int, int, int, int, int, int, int,
                       hipStream_t);
void dwconv_wgrad_bf16(const bf16*, const bf16*, float*, int, int, int,
                       int, int, int, int, int, int, int, int, int,
                       hipStream_t);
// gemm_nn.hip / attention entry points (gemm_tn.hip, gemm_wg.hip)
void gemm_nn_bf16(const bf16*, ZMap, long, int, const bf16*, ZMap, long,
                  int, bf16*, ZMap, long, int, int, const bf16*,
                  hipStream_t);
void attn_scores_bf16(const bf16*, const bf16*, bf16*, int, int, int, int,
                      int, const bf16*, hipStream_t);
void attn_wg_bf16(const bf16*, const bf16*, bf16*, int, int, int, int,
                  int, const bf16*, hipStream_t);
// flash.hip
void flash_fwd_bf16(const bf16*, const bf16*, const bf16*, bf16*, float*,
                    float*, int, int, int, int, int, float, bool,
                    const unsigned long long*, unsigned long long, float,
                    hipStream_t);
void flash_bwd_bf16(const bf16*, const bf16*, const bf16*, const bf16*,
                    const bf16*, const float*, const float*, float*,
                    float*, bf16*, bf16*, int, int, int, int, int, float,
                    bool, const unsigned long long*, unsigned long long,
                    float, hipStream_t);
void act_fwd_bf16(int, const bf16*, bf16*, long, hipStream_t);
void act_bwd_bf16(int, const bf16*, const bf16*, bf16*, long, hipStream_t);
// gemm_fp8.hip
void gemm_tn_fp8(const unsigned char*, const unsigned char*, bf16*,
                 const float*, const float*, const float*, int, int, int,
                 long, long, long, long, long, long, int, bool,
                 const unsigned char*, hipStream_t);
void gemm_mx_bf16(const unsigned char*, const unsigned char*,
                  const unsigned char*, const unsigned char*, bf16*,
                  const float*, int, int, int, long, int, int, int, float*,
                  const unsigned char*, hipStream_t);
int gemm_mx_nslots(int, int, int, int, int);
void rng_uniform_bf16(bf16*, long, const unsigned long long*,
                      unsigned long long, float, float, hipStream_t);
void rng_normal_bf16(bf16*, long, const unsigned long long*,
                     unsigned long long, float, float, hipStream_t);
// ew.hip (generic elementwise/broadcast/reduce launcher)
void ew_unary(int, const bf16*, bf16*, long, hipStream_t);
void ew_unary_grad(int, const bf16*, const bf16*, const bf16*, bf16*, long,
                   hipStream_t);
void ew_binary(int, const bf16*, const bf16*, bf16*, long, hipStream_t);
void ew_binary_bcast(int, const bf16*, const bf16*, bf16*, int, const long*,
                     const long*, const long*, long, hipStream_t);
void ew_reduce_last(int, const bf16*, float*, long, long, hipStream_t);
void gemm_tn_i8(const unsigned char*, const unsigned char*, int*, bf16*,
                const float*, const float*, int, int, int, long, long, long,
                const unsigned char*, hipStream_t);
void mx_cast_bf16(const bf16*, unsigned char*, unsigned char*, long,
                  hipStream_t);
void mx_cast_bf8(const bf16*, unsigned char*, unsigned char*, long,
                 hipStream_t);
void mx_cast_t(const bf16*, unsigned char*, unsigned char*, int, int, bool,
               hipStream_t);
void cast_to_fp8(const bf16*, unsigned char*, long, const float*,
                 hipStream_t);
// optim.hip
struct TensorDesc {
  void* w;
  void* g;
  float* m;
  float* v;
  float* w32;
  long n;
  long chunk0;
};
void multi_opt(const TensorDesc*, int, long, int, bool, float, float, float,
               float, float, float, float, float, hipStream_t);
void multi_sumsq(const TensorDesc*, int, long, bool, float*, hipStream_t);
}  // namespace mxa

namespace {

hipStream_t cur_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

const bf16* bfp(const at::Tensor& t) {
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, "expected bf16 tensor");
  return reinterpret_cast<const bf16*>(t.data_ptr());
}
bf16* bfp_mut(at::Tensor& t) {
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, "expected bf16 tensor");
  return reinterpret_cast<bf16*>(t.data_ptr());
}

// minimum 256^2-tile count for the deep-pipelined tn8 path (tunable)
static long tn8_minblocks() {
  static const long v = [] {
    const char* e = getenv("MXA_TN8_MINBLOCKS");
    return e ? atol(e) : 240L;
  }();
  return v;
}

// 128-byte zero page per device (OOB rows of tile staging read this)
at::Tensor& zpage_for(const at::Tensor& like) {
  static std::unordered_map<int, at::Tensor> cache;
  int dev = like.get_device();
  auto it = cache.find(dev);
  if (it == cache.end()) {
    auto z = at::zeros({128}, like.options().dtype(at::kBFloat16));
    it = cache.emplace(dev, z).first;
  }
  return it->second;
}

const bf16* zp(const at::Tensor& like) { return bfp(zpage_for(like)); }

}  // namespace

// ---------------------------------------------------------------------------
// GEMM / linear
// ---------------------------------------------------------------------------

// C[M,N] = A[M,K] @ B[N,K]^T (+bias) (+relu); 3-d inputs batch over dim 0.
static at::Tensor gemm_tn_impl(const at::Tensor& A, const at::Tensor& B,
                               c10::optional<at::Tensor> bias, bool relu,
                               float* stats) {
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous(), "gemm_tn: contiguous");
  bool batched = A.dim() == 3;
  long M = A.size(-2), K = A.size(-1), N = B.size(-2);
  TORCH_CHECK(B.size(-1) == K, "gemm_tn: K mismatch");
  TORCH_CHECK(K % 64 == 0, "gemm_tn: K must be a multiple of 64, got ", K);
  int batches = batched ? A.size(0) : 1;
  auto out = batched ? at::empty({batches, M, N}, A.options())
                     : at::empty({M, N}, A.options());
  const float* bptr = nullptr;
  at::Tensor bias_f;
  if (bias.has_value() && bias->defined()) {
    bias_f = bias->to(at::kFloat).contiguous();
    bptr = bias_f.data_ptr<float>();
  }
  long sA = batched ? M * K : 0;
  long sB = (batched && B.dim() == 3) ? N * K : 0;
  long sC = batched ? M * N : 0;
  long blocks8 = ((M + 255) / 256) * ((N + 255) / 256) * (long)batches;
  static const long tn8_mink = [] {
    const char* e = getenv("MXA_TN8_MINK");
    // short-K GEMMs fill better on the 128-tile kernel: tn8 runs 1
    // block/CU, and a K=768 FFN wgrad-shape grid of ~384 tiles loses a
    // third of the chip to the tail (562 vs 803 TF measured).  With the
    // 2-barrier-window kernel the crossover moved: K=1024 grids of
    // >=240 tiles now win on tn8 (835 vs 659 TF at 8192x4096x1024)
    return e ? atol(e) : 1024L;
  }();
  bool tn8_take = (blocks8 >= tn8_minblocks() && K >= tn8_mink) ||
                  (blocks8 >= 640 && K >= 768);  // huge grids: window
                  // kernel wins even at K=768 (744 vs 664 TF measured)
  static const bool tn8h_on = [] {
    const char* e = getenv("MXA_TN8H");
    return e && e[0] == '1';  // off until A/B'd
  }();
  long blocksh = ((M + 255) / 256) * ((N + 127) / 128) * (long)batches;
  if (M >= 256 && N >= 256 && tn8_take && !stats) {
    // deep-pipelined 256² 8-phase kernel (needs a full chip of tiles;
    // smaller grids run the 128² kernel which fills better)
    mxa::gemm_tn8_bf16(bfp(A), bfp(B), bfp_mut(out), bptr, (int)M, (int)N,
                       (int)K, K, K, N, sA, sB, sC, batches, relu, zp(A),
                       cur_stream());
  } else if (tn8h_on && M >= 256 && N >= 128 && K >= 768 &&
             blocksh >= 240 && !stats) {
    // 256x128 window variant: short-K grids the 256² tile can't fill
    mxa::gemm_tn8h_bf16(bfp(A), bfp(B), bfp_mut(out), bptr, (int)M,
                        (int)N, (int)K, K, K, N, sA, sB, sC, batches,
                        relu, zp(A), cur_stream());
  } else {
    // (measured: a split-K variant exists — gemm_tn_splitk_bf16 — but
    // auto-splitting these mid-size grids LOWERED throughput: the
    // atomic finish + shorter per-split pipelines cost more than the
    // extra fill buys.  Kept callable, not auto-dispatched.)
    mxa::gemm_tn_bf16(bfp(A), bfp(B), bfp_mut(out), bptr, (int)M, (int)N,
                      (int)K, K, K, N, sA, sB, sC, batches, relu, zp(A),
                      cur_stream(), stats);
  }
  return out;
}

at::Tensor gemm_tn(const at::Tensor& A, const at::Tensor& B,
                   c10::optional<at::Tensor> bias, bool relu) {
  return gemm_tn_impl(A, B, bias, relu, nullptr);
}

// gemm_tn with fused per-column BN statistics (the pointwise-conv -> BN
// fusion): returns {out} when the dispatch route cannot fill stats (the
// caller's BN then computes them itself), {out, stats[2,N]} when it can.
// tn8 wins by more than the stats pass saves on the deep-K shapes it
// owns, so stats are only produced on the 128-tile route when tn8 would
// not have been chosen anyway.
std::vector<at::Tensor> gemm_tn_stats(const at::Tensor& A,
                                      const at::Tensor& B,
                                      c10::optional<at::Tensor> bias,
                                      bool relu) {
  long M = A.size(-2), K = A.size(-1), N = B.size(-2);
  long blocks8 = ((M + 255) / 256) * ((N + 255) / 256);
  static const long tn8_mink2 = [] {
    const char* e = getenv("MXA_TN8_MINK");
    return e ? atol(e) : 1024L;
  }();
  if (A.dim() == 3 ||
      (M >= 256 && N >= 256 &&
       ((blocks8 >= tn8_minblocks() && K >= tn8_mink2) ||
        (blocks8 >= 640 && K >= 768))))
    return {gemm_tn_impl(A, B, bias, relu, nullptr)};
  auto stats = at::zeros({32, 2, N}, A.options().dtype(at::kFloat));
  auto out = gemm_tn_impl(A, B, bias, relu, stats.data_ptr<float>());
  return {out, stats};
}

// C[Mo,No] (bf16) = sum_m A[m,Mo] * B[m,No]; transpose -> C[No,Mo].
// Split-K goes through per-split f32 slot buffers folded (and cast, and
// optionally transposed) in one pass — no HBM atomics, no memset, no
// separate .to(bf16) launch.
at::Tensor gemm_wg(const at::Tensor& A, const at::Tensor& B,
                   bool transpose) {
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous(), "gemm_wg: contiguous");
  long m = A.size(0), Mo = A.size(1), No = B.size(1);
  TORCH_CHECK(B.size(0) == m, "gemm_wg: reduce dim mismatch");
  TORCH_CHECK(Mo % 8 == 0 && No % 8 == 0, "gemm_wg: dims must be %8");
  int ns = mxa::gemm_wg_nslots(m, (int)Mo, (int)No, transpose);
  auto out = transpose ? at::empty({No, Mo}, A.options())
                       : at::empty({Mo, No}, A.options());
  float* workp = nullptr;
  at::Tensor work;
  if (ns > 0) {
    work = at::empty({ns, Mo, No}, A.options().dtype(at::kFloat));
    workp = work.data_ptr<float>();
  }
  mxa::gemm_wg_bf16(bfp(A), bfp(B), workp, bfp_mut(out), transpose, m,
                    (int)Mo, (int)No, Mo, No, zp(A), cur_stream());
  return out;
}

// dW = A^T B with the bias gradient (colsum of A) folded into the same
// pass: returns {dW bf16, db f32[Mo]}.  Used by Linear/pointwise-conv
// backward where A is dy (reference: FullyConnectedGradCompute's
// separate bias-reduce in nn/fully_connected-inl.h).
std::vector<at::Tensor> gemm_wg_bias(const at::Tensor& A,
                                     const at::Tensor& B) {
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous(),
              "gemm_wg_bias: contiguous");
  long m = A.size(0), Mo = A.size(1), No = B.size(1);
  TORCH_CHECK(B.size(0) == m, "gemm_wg_bias: reduce dim mismatch");
  TORCH_CHECK(Mo % 8 == 0 && No % 8 == 0, "gemm_wg_bias: dims must be %8");
  int ns = mxa::gemm_wg_nslots(m, (int)Mo, (int)No, false);
  auto out = at::empty({Mo, No}, A.options());
  auto biasg = at::zeros({Mo}, A.options().dtype(at::kFloat));
  float* workp = nullptr;
  at::Tensor work;
  if (ns > 0) {
    work = at::empty({ns, Mo, No}, A.options().dtype(at::kFloat));
    workp = work.data_ptr<float>();
  }
  mxa::gemm_wg_bf16(bfp(A), bfp(B), workp, bfp_mut(out), false, m, (int)Mo,
                    (int)No, Mo, No, zp(A), cur_stream(),
                    biasg.data_ptr<float>());
  return {out, biasg};
}

// fp8 (e4m3) GEMM: A8 [.., M, K] u8, B8 [.., N, K] u8 -> bf16 C
at::Tensor gemm_tn_fp8(const at::Tensor& A8, const at::Tensor& B8,
                       c10::optional<at::Tensor> bias, const at::Tensor& sa,
                       const at::Tensor& sb, bool relu) {
  TORCH_CHECK(sa.scalar_type() == at::kFloat && sb.scalar_type() == at::kFloat);
  TORCH_CHECK(A8.scalar_type() == at::kByte && B8.scalar_type() == at::kByte);
  TORCH_CHECK(A8.is_contiguous() && B8.is_contiguous());
  bool batched = A8.dim() == 3;
  long M = A8.size(-2), K = A8.size(-1), N = B8.size(-2);
  TORCH_CHECK(B8.size(-1) == K && K % 128 == 0,
              "gemm_tn_fp8: K must be %128");
  int batches = batched ? A8.size(0) : 1;
  auto out = batched
                 ? at::empty({batches, M, N}, A8.options().dtype(at::kBFloat16))
                 : at::empty({M, N}, A8.options().dtype(at::kBFloat16));
  const float* bptr = nullptr;
  at::Tensor bias_f;
  if (bias.has_value() && bias->defined()) {
    bias_f = bias->to(at::kFloat).contiguous();
    bptr = bias_f.data_ptr<float>();
  }
  mxa::gemm_tn_fp8(A8.data_ptr<unsigned char>(),
                   B8.data_ptr<unsigned char>(), bfp_mut(out), bptr,
                   sa.data_ptr<float>(), sb.data_ptr<float>(),
                   (int)M, (int)N, (int)K, K, K, N,
                   batched ? M * K : 0,
                   (batched && B8.dim() == 3) ? N * K : 0,
                   batched ? M * N : 0, batches, relu,
                   (const unsigned char*)zp(zpage_for(out)), cur_stream());
  return out;
}

std::vector<at::Tensor> mx_cast(const at::Tensor& x, bool e5) {
  // bf16 [.., K] -> (fp8 payload u8, e8m0 block scales u8 [.., K/32]);
  // e5 -> e5m2 (gradients), else e4m3
  TORCH_CHECK(x.is_contiguous() && x.size(-1) % 32 == 0,
              "mx_cast: last dim must be %32");
  long nblk = x.numel() / 32;
  auto q = at::empty(x.sizes(), x.options().dtype(at::kByte));
  auto shp = x.sizes().vec();
  shp.back() /= 32;
  auto sc = at::empty(shp, x.options().dtype(at::kByte));
  if (e5)
    mxa::mx_cast_bf8(bfp(x), q.data_ptr<unsigned char>(),
                     sc.data_ptr<unsigned char>(), nblk, cur_stream());
  else
    mxa::mx_cast_bf16(bfp(x), q.data_ptr<unsigned char>(),
                      sc.data_ptr<unsigned char>(), nblk, cur_stream());
  return {q, sc};
}

std::vector<at::Tensor> mx_cast_t(const at::Tensor& x, bool e5) {
  // bf16 [R, C] -> transposed MX: (Q u8 [C, R], SC u8 [C, R/32]),
  // quantization blocks of 32 along R — grads/operands for the backward
  // GEMMs expressed as plain TN MX GEMMs over transposed operands
  TORCH_CHECK(x.is_contiguous() && x.dim() == 2 && x.size(0) % 32 == 0,
              "mx_cast_t: [R, C] with R % 32 == 0");
  int R = (int)x.size(0), C = (int)x.size(1);
  auto q = at::empty({(long)C, (long)R}, x.options().dtype(at::kByte));
  auto sc = at::empty({(long)C, (long)R / 32},
                      x.options().dtype(at::kByte));
  mxa::mx_cast_t(bfp(x), q.data_ptr<unsigned char>(),
                 sc.data_ptr<unsigned char>(), R, C, e5, cur_stream());
  return {q, sc};
}

at::Tensor gemm_mx(const at::Tensor& A8, const at::Tensor& Asc,
                   const at::Tensor& B8, const at::Tensor& Bsc,
                   const c10::optional<at::Tensor>& bias, bool relu,
                   long afmt, long bfmt) {
  // C[M][N] bf16 = MX-fp8 A[M,K] @ B[N,K]^T (both k-contiguous)
  TORCH_CHECK(A8.dim() == 2 && B8.dim() == 2);
  long M = A8.size(0), K = A8.size(1), N = B8.size(0);
  TORCH_CHECK(B8.size(1) == K && K % 128 == 0, "gemm_mx: K % 128 != 0");
  auto opts = at::TensorOptions().device(A8.device()).dtype(at::kBFloat16);
  auto out = at::empty({M, N}, opts);
  auto zprobe = at::empty({0}, opts);
  int ns = mxa::gemm_mx_nslots((int)M, (int)N, (int)K,
                               bias.has_value() ? 1 : 0, relu ? 1 : 0);
  at::Tensor work;
  float* wp = nullptr;
  if (ns > 0) {
    work = at::empty({(long)ns, M, N}, opts.dtype(at::kFloat));
    wp = work.data_ptr<float>();
  }
  mxa::gemm_mx_bf16(A8.data_ptr<unsigned char>(),
                    Asc.data_ptr<unsigned char>(),
                    B8.data_ptr<unsigned char>(),
                    Bsc.data_ptr<unsigned char>(), bfp_mut(out),
                    bias.has_value() ? bias->data_ptr<float>() : nullptr,
                    (int)M, (int)N, (int)K, N, relu, (int)afmt,
                    (int)bfmt, wp, (const unsigned char*)zp(zprobe),
                    cur_stream());
  return out;
}

// int8 TN GEMM (quantized FC / pointwise conv compute).  inv_scale none
// -> exact int32 accumulator out; else fused dequant to bf16 (+bias).
at::Tensor gemm_i8(const at::Tensor& A8, const at::Tensor& B8,
                   const c10::optional<at::Tensor>& inv_scale,
                   const c10::optional<at::Tensor>& bias) {
  TORCH_CHECK(A8.scalar_type() == at::kChar && B8.scalar_type() == at::kChar,
              "gemm_i8: int8 operands");
  TORCH_CHECK(A8.is_contiguous() && B8.is_contiguous());
  long M = A8.size(0), K = A8.size(1), N = B8.size(0);
  TORCH_CHECK(B8.size(1) == K && K % 128 == 0, "gemm_i8: K % 128 != 0");
  auto zprobe = at::empty(
      {0}, at::TensorOptions().device(A8.device()).dtype(at::kBFloat16));
  if (!inv_scale.has_value()) {
    auto out = at::empty({M, N},
                         A8.options().dtype(at::kInt));
    mxa::gemm_tn_i8((const unsigned char*)A8.data_ptr<signed char>(),
                    (const unsigned char*)B8.data_ptr<signed char>(),
                    out.data_ptr<int>(), nullptr, nullptr, nullptr, (int)M,
                    (int)N, (int)K, K, K, N,
                    (const unsigned char*)zp(zprobe), cur_stream());
    return out;
  }
  auto out = at::empty({M, N}, A8.options().dtype(at::kBFloat16));
  mxa::gemm_tn_i8((const unsigned char*)A8.data_ptr<signed char>(),
                  (const unsigned char*)B8.data_ptr<signed char>(), nullptr,
                  bfp_mut(out), inv_scale->data_ptr<float>(),
                  bias.has_value() ? bias->data_ptr<float>() : nullptr,
                  (int)M, (int)N, (int)K, K, K, N,
                  (const unsigned char*)zp(zprobe), cur_stream());
  return out;
}

// ---- generic elementwise launcher (ew.hip) ----
at::Tensor& rng_seed_for(const at::Tensor& like);  // defined below

at::Tensor rng_sample(std::vector<long> shape, double a, double b,
                      long salt, bool normal, long device) {
  auto opts = at::TensorOptions()
                  .device(at::Device(at::kCUDA, (int)device))
                  .dtype(at::kBFloat16);
  long n = 1;
  for (long d : shape) n *= d;
  long npad = (n + 7) / 8 * 8;
  auto out = at::empty({npad}, opts);
  auto& sw = rng_seed_for(out);
  if (normal)
    mxa::rng_normal_bf16(bfp_mut(out), npad,
                         (const unsigned long long*)sw.data_ptr<long>(),
                         (unsigned long long)salt, (float)a, (float)b,
                         cur_stream());
  else
    mxa::rng_uniform_bf16(bfp_mut(out), npad,
                          (const unsigned long long*)sw.data_ptr<long>(),
                          (unsigned long long)salt, (float)a, (float)b,
                          cur_stream());
  return out.narrow(0, 0, n).reshape(shape);
}

at::Tensor ew_unary_rt(long op, const at::Tensor& x) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  auto y = at::empty_like(x);
  mxa::ew_unary((int)op, bfp(x), bfp_mut(y), x.numel(), cur_stream());
  return y;
}

at::Tensor ew_unary_grad_rt(long op, const at::Tensor& x,
                            const at::Tensor& y, const at::Tensor& dy) {
  auto dx = at::empty_like(x);
  mxa::ew_unary_grad((int)op, bfp(x), bfp(y), bfp(dy), bfp_mut(dx),
                     x.numel(), cur_stream());
  return dx;
}

at::Tensor ew_binary_rt(long op, const at::Tensor& a, const at::Tensor& b) {
  TORCH_CHECK(a.scalar_type() == at::kBFloat16 &&
              b.scalar_type() == at::kBFloat16);
  if (a.sizes() == b.sizes() && a.is_contiguous() && b.is_contiguous()) {
    auto y = at::empty_like(a);
    mxa::ew_binary((int)op, bfp(a), bfp(b), bfp_mut(y), a.numel(),
                   cur_stream());
    return y;
  }
  // broadcast: align shapes right, size-1 dims get stride 0
  auto out_sizes = at::infer_size(a.sizes(), b.sizes());
  int nd = (int)out_sizes.size();
  TORCH_CHECK(nd <= 6, "ew_binary: ndim > 6");
  long shape[6], sa[6], sb[6];
  auto aa = a, bb = b;
  while (aa.dim() < nd) aa = aa.unsqueeze(0);
  while (bb.dim() < nd) bb = bb.unsqueeze(0);
  aa = aa.contiguous();
  bb = bb.contiguous();
  long n = 1;
  for (int d = 0; d < nd; ++d) {
    shape[d] = out_sizes[d];
    n *= out_sizes[d];
    sa[d] = aa.size(d) == 1 ? 0 : aa.stride(d);
    sb[d] = bb.size(d) == 1 ? 0 : bb.stride(d);
    TORCH_CHECK(aa.size(d) == 1 || aa.size(d) == out_sizes[d]);
    TORCH_CHECK(bb.size(d) == 1 || bb.size(d) == out_sizes[d]);
  }
  auto y = at::empty(out_sizes, a.options());
  mxa::ew_binary_bcast((int)op, bfp(aa), bfp(bb), bfp_mut(y), nd, shape,
                       sa, sb, n, cur_stream());
  return y;
}

at::Tensor ew_reduce_last_rt(long op, const at::Tensor& x) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_contiguous() &&
              x.dim() >= 1);
  long D = x.size(-1);
  long rows = x.numel() / D;
  auto out = at::empty({rows}, x.options().dtype(at::kFloat));
  mxa::ew_reduce_last((int)op, bfp(x), out.data_ptr<float>(), rows, D,
                      cur_stream());
  auto shp = x.sizes().vec();
  shp.pop_back();
  return shp.empty() ? out.reshape({}) : out.reshape(shp);
}

at::Tensor cast_to_fp8(const at::Tensor& x, const at::Tensor& scale) {
  long n = x.numel();
  TORCH_CHECK(n % 8 == 0);
  TORCH_CHECK(scale.scalar_type() == at::kFloat);
  auto y = at::empty_like(x, x.options().dtype(at::kByte));
  mxa::cast_to_fp8(bfp(x), y.data_ptr<unsigned char>(), n,
                   scale.data_ptr<float>(), cur_stream());
  return y;
}

// ---------------------------------------------------------------------------
// Convolution (NHWC)
// ---------------------------------------------------------------------------

static at::Tensor conv2d_fwd_impl(const at::Tensor& x,
                                  const at::Tensor& w_rskc,
                                  c10::optional<at::Tensor> bias,
                                  std::vector<long> stride,
                                  std::vector<long> pad,
                                  std::vector<long> dil, bool relu,
                                  float* stats, bool* stats_filled) {
  TORCH_CHECK(x.is_contiguous() && w_rskc.is_contiguous());
  int Nb = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int R = w_rskc.size(0), S = w_rskc.size(1), Kout = w_rskc.size(2);
  TORCH_CHECK(w_rskc.size(3) == C, "conv2d_fwd: channel mismatch");
  TORCH_CHECK(C % 64 == 0, "conv2d_fwd: C must be %64 (stem uses im2col)");
  int sh = stride[0], sw = stride[1], ph = pad[0], pw = pad[1],
      dh = dil[0], dw = dil[1];
  int P = (H + 2 * ph - dh * (R - 1) - 1) / sh + 1;
  int Q = (W + 2 * pw - dw * (S - 1) - 1) / sw + 1;
  auto y = at::empty({Nb, P, Q, Kout}, x.options());
  const float* bptr = nullptr;
  at::Tensor bias_f;
  if (bias.has_value() && bias->defined()) {
    bias_f = bias->to(at::kFloat).contiguous();
    bptr = bias_f.data_ptr<float>();
  }
  long M = (long)Nb * P * Q;
  long blocks = ((M + 127) / 128) * ((Kout + 127) / 128);
  at::Tensor accbuf;
  float* accptr = nullptr;
  if (blocks < 150 && R * S * (C / 64) > 1) {
    accbuf = at::empty({M, Kout}, x.options().dtype(at::kFloat));
    accptr = accbuf.data_ptr<float>();
  }
  bool filled = mxa::conv2d_fwd_bf16(
      bfp(x), bfp(w_rskc), bfp_mut(y), accptr, bptr, Nb, H, W, C, Kout, R,
      S, P, Q, sh, sw, ph, pw, dh, dw, relu, zp(x), cur_stream(), stats);
  if (stats_filled) *stats_filled = filled;
  return y;
}

at::Tensor conv2d_fwd(const at::Tensor& x, const at::Tensor& w_rskc,
                      c10::optional<at::Tensor> bias,
                      std::vector<long> stride, std::vector<long> pad,
                      std::vector<long> dil, bool relu) {
  return conv2d_fwd_impl(x, w_rskc, bias, stride, pad, dil, relu, nullptr,
                         nullptr);
}

// conv2d_fwd with fused per-channel BN statistics accumulated in the GEMM
// epilogue: {y, stats[2,K]} when the unsplit 128-tile route ran, {y} when
// the split-K path was taken (its finish kernel does not fold stats; the
// caller's BN computes them itself).
std::vector<at::Tensor> conv2d_fwd_stats(const at::Tensor& x,
                                         const at::Tensor& w_rskc,
                                         c10::optional<at::Tensor> bias,
                                         std::vector<long> stride,
                                         std::vector<long> pad,
                                         std::vector<long> dil, bool relu) {
  int Kout = w_rskc.size(2);
  auto stats = at::zeros({32, 2, Kout}, x.options().dtype(at::kFloat));
  bool filled = false;
  auto y = conv2d_fwd_impl(x, w_rskc, bias, stride, pad, dil, relu,
                           stats.data_ptr<float>(), &filled);
  if (!filled) return {y};
  return {y, stats};
}

at::Tensor conv2d_dgrad(const at::Tensor& dy, const at::Tensor& w_rsck,
                        std::vector<long> xshape, std::vector<long> stride,
                        std::vector<long> pad, std::vector<long> dil) {
  TORCH_CHECK(dy.is_contiguous() && w_rsck.is_contiguous());
  int Nb = xshape[0], H = xshape[1], W = xshape[2], C = xshape[3];
  int R = w_rsck.size(0), S = w_rsck.size(1);
  TORCH_CHECK(w_rsck.size(2) == C, "conv2d_dgrad: channel mismatch");
  int Kout = w_rsck.size(3);
  TORCH_CHECK(Kout % 64 == 0, "conv2d_dgrad: K must be %64");
  int P = dy.size(1), Q = dy.size(2);
  auto dx = at::empty({Nb, H, W, C}, dy.options());
  long M = (long)Nb * H * W;
  long blocks = ((M + 127) / 128) * ((C + 127) / 128);
  at::Tensor accbuf;
  float* accptr = nullptr;
  if (blocks < 150 && R * S * (Kout / 64) > 1) {
    accbuf = at::empty({M, C}, dy.options().dtype(at::kFloat));
    accptr = accbuf.data_ptr<float>();
  }
  mxa::conv2d_dgrad_bf16(bfp(dy), bfp(w_rsck), bfp_mut(dx), accptr, Nb, H,
                         W, C, Kout, R, S, P, Q, (int)stride[0],
                         (int)stride[1], (int)pad[0], (int)pad[1],
                         (int)dil[0], (int)dil[1], zp(dy), cur_stream());
  return dx;
}

// Returns dW in (K, R, S, C) bf16 directly: the split slots are folded,
// cast and transposed in one epilogue pass (the old path emitted f32
// [R,S,C,K] + a torch permute + a torch cast per conv per step).
at::Tensor conv2d_wgrad(const at::Tensor& x, const at::Tensor& dy,
                        std::vector<long> kernel, std::vector<long> stride,
                        std::vector<long> pad, std::vector<long> dil) {
  TORCH_CHECK(x.is_contiguous() && dy.is_contiguous());
  int Nb = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int P = dy.size(1), Q = dy.size(2), Kout = dy.size(3);
  int R = kernel[0], S = kernel[1];
  static const bool c9_on = [] {
    const char* e = getenv("MXA_WG_C9");
    return !e || e[0] != '0';  // default on; MXA_WG_C9=0 -> per-seg path
  }();
  if (c9_on &&
      mxa::conv_wg_c9_eligible(R, S, (int)stride[0], (int)stride[1],
                               (int)pad[0], (int)pad[1], (int)dil[0],
                               (int)dil[1], H, W, P, Q, C, Kout)) {
    // seg-merged path: all 9 (r,s) tiles from one staged stream of a
    // padded copy (see gemm_wg.hip "c9"); x+dy read once instead of 9x
    auto xp = at::empty({mxa::conv_wg_c9_xp_elems(Nb, H, W, C)},
                        x.options());
    mxa::conv_wg_pad_image(bfp(x), bfp_mut(xp), Nb, H, W, C, cur_stream());
    int ns = mxa::conv2d_wgrad_c9_nslots(Nb, H, W, C, Kout);
    auto work = at::empty({ns, 9L * C, (long)Kout},
                          x.options().dtype(at::kFloat));
    auto dw = at::empty({Kout, 3, 3, C}, x.options());
    mxa::conv2d_wgrad_c9_bf16(bfp(xp), bfp(dy), work.data_ptr<float>(),
                              bfp_mut(dw), Nb, H, W, C, Kout, zp(x),
                              cur_stream());
    return dw;
  }
  long mtotal = (long)Nb * P * Q;
  int ns = mxa::conv2d_wgrad_nslots(mtotal, C, Kout, R, S);
  auto work = at::empty({ns, (long)R * S * C, (long)Kout},
                        x.options().dtype(at::kFloat));
  auto dw = at::empty({Kout, R, S, C}, x.options());
  mxa::conv2d_wgrad_bf16(bfp(x), bfp(dy), work.data_ptr<float>(),
                         bfp_mut(dw), Nb, H, W, C, Kout, R, S, P, Q,
                         (int)stride[0], (int)stride[1], (int)pad[0],
                         (int)pad[1], (int)dil[0], (int)dil[1], zp(x),
                         cur_stream());
  return dw;
}

// ---------------------------------------------------------------------------
// BatchNorm (NHWC) / pooling
// ---------------------------------------------------------------------------

std::vector<at::Tensor> bn_fwd_train(const at::Tensor& x,
                                     const at::Tensor& gamma,
                                     const at::Tensor& beta,
                                     at::Tensor running_mean,
                                     at::Tensor running_var, double eps,
                                     double momentum, bool relu,
                                     const c10::optional<at::Tensor>& resid,
                                     const c10::optional<at::Tensor>& psums) {
  int C = x.size(-1);
  long M = x.numel() / C;
  auto y = at::empty_like(x);
  auto opts = x.options().dtype(at::kFloat);
  auto acc = at::empty({2, C}, opts);   // [psum; psumsq], one memset
  auto psum = acc[0], psumsq = acc[1];
  auto save_mean = at::empty({C}, opts), save_rstd = at::empty({C}, opts);
  const float* ext = nullptr;
  if (psums.has_value() && psums->defined()) {
    TORCH_CHECK(psums->is_contiguous() && psums->numel() == 64 * C &&
                    psums->scalar_type() == at::kFloat,
                "bn_fwd_train: psums must be contiguous f32 [32,2,C]");
    ext = psums->data_ptr<float>();
  }
  mxa::bn_fwd_train(bfp(x), bfp_mut(y), psum.data_ptr<float>(),
                    psumsq.data_ptr<float>(), save_mean.data_ptr<float>(),
                    save_rstd.data_ptr<float>(),
                    running_mean.data_ptr<float>(),
                    running_var.data_ptr<float>(), gamma.data_ptr<float>(),
                    beta.data_ptr<float>(), M, C, (float)eps,
                    (float)momentum, relu,
                    resid.has_value() ? bfp(*resid) : nullptr,
                    cur_stream(), ext);
  return {y, save_mean, save_rstd};
}

at::Tensor bn_fwd_infer(const at::Tensor& x, const at::Tensor& gamma,
                        const at::Tensor& beta, const at::Tensor& rmean,
                        const at::Tensor& rvar, double eps, bool relu,
                        const c10::optional<at::Tensor>& resid) {
  int C = x.size(-1);
  long M = x.numel() / C;
  auto y = at::empty_like(x);
  mxa::bn_fwd_infer(bfp(x), bfp_mut(y), rmean.data_ptr<float>(),
                    rvar.data_ptr<float>(), gamma.data_ptr<float>(),
                    beta.data_ptr<float>(), M, C, (float)eps, relu,
                    resid.has_value() ? bfp(*resid) : nullptr,
                    cur_stream());
  return y;
}

std::vector<at::Tensor> bn_bwd(const at::Tensor& x, const at::Tensor& dy,
                               const at::Tensor& gamma,
                               const at::Tensor& beta,
                               const at::Tensor& save_mean,
                               const at::Tensor& save_rstd, bool relu) {
  int C = x.size(-1);
  long M = x.numel() / C;
  auto dx = at::empty_like(x);
  auto opts = x.options().dtype(at::kFloat);
  auto dacc = at::empty({2, C}, opts);  // [dbeta; dgamma], one memset
  auto dbeta = dacc[0], dgamma = dacc[1];
  mxa::bn_bwd(bfp(x), bfp(dy), bfp_mut(dx), save_mean.data_ptr<float>(),
              save_rstd.data_ptr<float>(), gamma.data_ptr<float>(),
              beta.data_ptr<float>(), dgamma.data_ptr<float>(),
              dbeta.data_ptr<float>(), M, C, relu, cur_stream());
  return {dx, dgamma, dbeta};
}

std::vector<at::Tensor> maxpool(const at::Tensor& x,
                                std::vector<long> kernel,
                                std::vector<long> stride,
                                std::vector<long> pad) {
  int Nb = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int R = kernel[0], S = kernel[1], sh = stride[0], sw = stride[1],
      ph = pad[0], pw = pad[1];
  int P = (H + 2 * ph - R) / sh + 1;
  int Q = (W + 2 * pw - S) / sw + 1;
  auto y = at::empty({Nb, P, Q, C}, x.options());
  auto arg = at::empty({Nb, P, Q, C}, x.options().dtype(at::kByte));
  mxa::maxpool_nhwc(bfp(x), bfp_mut(y), arg.data_ptr<unsigned char>(), Nb,
                    H, W, C, P, Q, R, S, sh, sw, ph, pw, cur_stream());
  return {y, arg};
}

at::Tensor maxpool_bwd(const at::Tensor& dy, const at::Tensor& arg,
                       std::vector<long> xshape, std::vector<long> kernel,
                       std::vector<long> stride, std::vector<long> pad) {
  int Nb = xshape[0], H = xshape[1], W = xshape[2], C = xshape[3];
  int P = dy.size(1), Q = dy.size(2);
  auto dx = at::empty({Nb, H, W, C}, dy.options());
  mxa::maxpool_nhwc_bwd(bfp(dy), arg.data_ptr<unsigned char>(),
                        bfp_mut(dx), Nb, H, W, C, P, Q,
                        (int)kernel[0], (int)kernel[1], (int)stride[0],
                        (int)stride[1], (int)pad[0], (int)pad[1],
                        cur_stream());
  return dx;
}

at::Tensor avgpool(const at::Tensor& x, std::vector<long> kernel,
                   std::vector<long> stride, std::vector<long> pad,
                   bool count_pad) {
  int Nb = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int R = kernel[0], S = kernel[1], sh = stride[0], sw = stride[1],
      ph = pad[0], pw = pad[1];
  int P = (H + 2 * ph - R) / sh + 1;
  int Q = (W + 2 * pw - S) / sw + 1;
  auto y = at::empty({Nb, P, Q, C}, x.options());
  mxa::avgpool_nhwc(bfp(x), bfp_mut(y), Nb, H, W, C, P, Q, R, S, sh, sw,
                    ph, pw, count_pad, cur_stream());
  return y;
}

at::Tensor global_avgpool(const at::Tensor& x) {
  int Nb = x.size(0), C = x.size(3);
  long HW = (long)x.size(1) * x.size(2);
  auto y = at::empty({Nb, 1, 1, C}, x.options());
  mxa::gap_nhwc(bfp(x), bfp_mut(y), Nb, HW, C, cur_stream());
  return y;
}

at::Tensor global_avgpool_bwd(const at::Tensor& dy,
                              std::vector<long> xshape) {
  int Nb = xshape[0], C = xshape[3];
  long HW = (long)xshape[1] * xshape[2];
  auto dx = at::empty({Nb, xshape[1], xshape[2], C}, dy.options());
  mxa::gap_nhwc_bwd(bfp(dy), bfp_mut(dx), Nb, HW, C, cur_stream());
  return dx;
}

// ---------------------------------------------------------------------------
// softmax / layernorm / dropout / embedding / activations
// ---------------------------------------------------------------------------

at::Tensor softmax_fwd(const at::Tensor& x, bool log_, double alpha) {
  int D = x.size(-1);
  TORCH_CHECK(D % 8 == 0, "softmax: last dim must be %8");
  long rows = x.numel() / D;
  auto y = at::empty_like(x);
  mxa::softmax_fwd_bf16(bfp(x), bfp_mut(y), rows, D, log_, (float)alpha,
                        cur_stream());
  return y;
}

at::Tensor softmax_masked_fwd(const at::Tensor& x, const at::Tensor& vlen,
                              double alpha) {
  int D = x.size(-1);
  TORCH_CHECK(D % 8 == 0, "softmax: last dim must be %8");
  long rows = x.numel() / D;
  TORCH_CHECK(vlen.numel() == rows && vlen.scalar_type() == at::kInt);
  auto y = at::empty_like(x);
  mxa::softmax_masked_fwd_bf16(bfp(x), bfp_mut(y), rows, D,
                               vlen.data_ptr<int>(), (float)alpha,
                               cur_stream());
  return y;
}

at::Tensor softmax_bwd(const at::Tensor& y, const at::Tensor& dy,
                       bool log_, double alpha) {
  int D = y.size(-1);
  long rows = y.numel() / D;
  auto dx = at::empty_like(y);
  mxa::softmax_bwd_bf16(bfp(y), bfp(dy), bfp_mut(dx), rows, D, log_,
                        (float)alpha, cur_stream());
  return dx;
}

// persistent device RNG seed word per device (graph-safe dropout)
at::Tensor& rng_seed_for(const at::Tensor& like) {
  static std::unordered_map<int, at::Tensor> cache;
  int dev = like.get_device();
  auto it = cache.find(dev);
  if (it == cache.end()) {
    auto z = at::full({1}, (long)0x243F6A8885A308D3LL,
                      like.options().dtype(at::kLong));
    it = cache.emplace(dev, z).first;
  }
  return it->second;
}

std::vector<at::Tensor> softmax_dropout_fwd(
    const at::Tensor& x, const c10::optional<at::Tensor>& vlen,
    double alpha, double pdrop, long salt) {
  int D = x.size(-1);
  TORCH_CHECK(D % 8 == 0, "softmax_dropout: last dim must be %8");
  long rows = x.numel() / D;
  auto p = at::empty_like(x);
  auto pd = at::empty_like(x);
  auto mask = at::empty({x.numel()}, x.options().dtype(at::kByte));
  auto& sw = rng_seed_for(x);
  mxa::softmax_dropout_fwd_bf16(
      bfp(x), bfp_mut(p), bfp_mut(pd), mask.data_ptr<unsigned char>(),
      rows, D, vlen.has_value() ? vlen->data_ptr<int>() : nullptr,
      (float)alpha, (const unsigned long long*)sw.data_ptr<long>(),
      (unsigned long long)salt, (float)pdrop, cur_stream());
  return {p, pd, mask};
}

at::Tensor softmax_dropout_bwd(const at::Tensor& p, const at::Tensor& dpd,
                               const at::Tensor& mask, double alpha,
                               double pdrop) {
  int D = p.size(-1);
  long rows = p.numel() / D;
  auto ds = at::empty_like(p);
  mxa::softmax_dropout_bwd_bf16(bfp(p), bfp(dpd),
                                mask.data_ptr<unsigned char>(),
                                bfp_mut(ds), rows, D, (float)alpha,
                                (float)pdrop, cur_stream());
  return ds;
}

std::vector<at::Tensor> layernorm_fwd(const at::Tensor& x,
                                      const at::Tensor& gamma,
                                      const at::Tensor& beta, double eps) {
  int D = x.size(-1);
  TORCH_CHECK(D % 8 == 0, "layernorm: last dim must be %8");
  long rows = x.numel() / D;
  auto y = at::empty_like(x);
  auto opts = x.options().dtype(at::kFloat);
  auto mean = at::empty({rows}, opts), rstd = at::empty({rows}, opts);
  mxa::layernorm_fwd_bf16(bfp(x), bfp_mut(y), gamma.data_ptr<float>(),
                          beta.data_ptr<float>(), mean.data_ptr<float>(),
                          rstd.data_ptr<float>(), rows, D, (float)eps,
                          cur_stream());
  return {y, mean, rstd};
}

std::vector<at::Tensor> layernorm_bwd(const at::Tensor& x,
                                      const at::Tensor& dy,
                                      const at::Tensor& gamma,
                                      const at::Tensor& mean,
                                      const at::Tensor& rstd) {
  int D = x.size(-1);
  long rows = x.numel() / D;
  auto dx = at::empty_like(x);
  auto opts = x.options().dtype(at::kFloat);
  auto dgamma = at::empty({D}, opts), dbeta = at::empty({D}, opts);
  mxa::layernorm_bwd_bf16(bfp(x), bfp(dy), bfp_mut(dx),
                          gamma.data_ptr<float>(), mean.data_ptr<float>(),
                          rstd.data_ptr<float>(), dgamma.data_ptr<float>(),
                          dbeta.data_ptr<float>(), rows, D, cur_stream());
  return {dx, dgamma, dbeta};
}

at::Tensor gemm_nn(const at::Tensor& a, const at::Tensor& b) {
  // C[M][N] = A[M][k] @ B[k][N], A k-contiguous, B out-contiguous — the
  // no-transpose dgrad form (dx = dy @ W without W^T copy).
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.is_contiguous() &&
              b.is_contiguous());
  long M = a.size(0), K = a.size(1), N = b.size(1);
  TORCH_CHECK(b.size(0) == K && K % 64 == 0 && N % 8 == 0,
              "gemm_nn: K%64, N%8 required");
  auto out = at::empty({M, N}, a.options());
  mxa::gemm_nn_bf16(bfp(a), ZMap{1, 0, 0}, K, (int)M, bfp(b),
                    ZMap{1, 0, 0}, N, (int)N, bfp_mut(out),
                    ZMap{1, 0, 0}, N, (int)K, 1, zp(a), cur_stream());
  return out;
}

at::Tensor attn_scores(const at::Tensor& q, const at::Tensor& k, long H) {
  // S[b*H+h][tq][tk] = sum_d q[b,tq,h*D+d] k[b,tk,h*D+d] — heads read in
  // place from [B,S,H*D] (no permute copies).  Also dP with (dO, V).
  TORCH_CHECK(q.dim() == 3 && k.dim() == 3 && q.is_contiguous() &&
              k.is_contiguous());
  long B = q.size(0), Tq = q.size(1), HD = q.size(2), Tk = k.size(1);
  long D = HD / H;
  TORCH_CHECK(D * H == HD && D % 64 == 0, "attn_scores: D%64 required");
  auto out = at::empty({B * H, Tq, Tk}, q.options());
  mxa::attn_scores_bf16(bfp(q), bfp(k), bfp_mut(out), (int)B, (int)H,
                        (int)Tq, (int)Tk, (int)D, zp(q), cur_stream());
  return out;
}

at::Tensor attn_ctx(const at::Tensor& p, const at::Tensor& v, long H) {
  // O[b,tq,h*D+d] = sum_tk P[b*H+h][tq][tk] v[b,tk,h*D+d] (NN form,
  // strided head output written in place).  Also dQ with (dS, K).
  TORCH_CHECK(p.dim() == 3 && v.dim() == 3 && p.is_contiguous() &&
              v.is_contiguous());
  long BH = p.size(0), Tq = p.size(1), Tk = p.size(2);
  long B = BH / H, HD = v.size(2), D = HD / H;
  TORCH_CHECK(B * H == BH && D * H == HD && Tk % 64 == 0 && D % 8 == 0,
              "attn_ctx: Tk%64, D%8 required");
  auto out = at::empty({B, Tq, HD}, p.options());
  mxa::gemm_nn_bf16(bfp(p), ZMap{1, Tq * Tk, 0}, Tk, (int)Tq, bfp(v),
                    ZMap{(int)H, Tk * HD, D}, HD, (int)D,
                    bfp_mut(out), ZMap{(int)H, Tq * HD, D}, HD,
                    (int)Tk, (int)BH, zp(p), cur_stream());
  return out;
}

std::vector<at::Tensor> flash_fwd(const at::Tensor& q,
                                  const at::Tensor& k,
                                  const at::Tensor& v, long H,
                                  double alpha, bool causal, double pdrop,
                                  long salt) {
  // q/k/v [B, T, H*D] bf16, D == 64, any Tq/Tk (tail-masked in-kernel)
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  long B = q.size(0), Tq = q.size(1), HD = q.size(2), Tk = k.size(1);
  long D = HD / H;
  TORCH_CHECK((D == 64 || D == 128) && Tq >= 1 && Tk >= 1,
              "flash_fwd: D in {64, 128} required");
  auto out = at::empty_like(q);
  auto opts = q.options().dtype(at::kFloat);
  auto m = at::empty({B * H, Tq}, opts);
  auto l = at::empty({B * H, Tq}, opts);
  auto& sw = rng_seed_for(q);
  mxa::flash_fwd_bf16(bfp(q), bfp(k), bfp(v), bfp_mut(out),
                      m.data_ptr<float>(), l.data_ptr<float>(), (int)B,
                      (int)H, (int)Tq, (int)Tk, (int)D, (float)alpha,
                      causal,
                      (const unsigned long long*)sw.data_ptr<long>(),
                      (unsigned long long)salt, (float)pdrop,
                      cur_stream());
  return {out, m, l};
}

std::vector<at::Tensor> flash_bwd(const at::Tensor& q,
                                  const at::Tensor& k,
                                  const at::Tensor& v,
                                  const at::Tensor& o,
                                  const at::Tensor& dout,
                                  const at::Tensor& m, const at::Tensor& l,
                                  long H, double alpha, bool causal,
                                  double pdrop, long salt) {
  long B = q.size(0), Tq = q.size(1), HD = q.size(2), Tk = k.size(1);
  auto opts = q.options().dtype(at::kFloat);
  auto drow = at::empty({B * H, Tq}, opts);
  auto dqacc = at::empty({B, Tq, HD}, opts);  // zeroed inside flash_drow
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  auto& sw = rng_seed_for(q);
  mxa::flash_bwd_bf16(bfp(q), bfp(k), bfp(v), bfp(o),
                      bfp(dout.contiguous()), m.data_ptr<float>(),
                      l.data_ptr<float>(), drow.data_ptr<float>(),
                      dqacc.data_ptr<float>(), bfp_mut(dk), bfp_mut(dv),
                      (int)B, (int)H, (int)Tq, (int)Tk, (int)(HD / H),
                      (float)alpha, causal,
                      (const unsigned long long*)sw.data_ptr<long>(),
                      (unsigned long long)salt, (float)pdrop,
                      cur_stream());
  return {dqacc.to(at::kBFloat16), dk, dv};
}

at::Tensor attn_wg(const at::Tensor& p, const at::Tensor& q, long H) {
  // dK[b,tk,h*D+d] = sum_tq dS[b*H+h][tq][tk] q[b,tq,h*D+d] (WG form,
  // f32 out in [B,S,H*D] layout).  Also dV with (P, dO).
  TORCH_CHECK(p.dim() == 3 && q.dim() == 3 && p.is_contiguous() &&
              q.is_contiguous());
  long BH = p.size(0), Tq = p.size(1), Tk = p.size(2);
  long B = BH / H, HD = q.size(2), D = HD / H;
  TORCH_CHECK(B * H == BH && D * H == HD && Tk % 8 == 0 && D % 8 == 0,
              "attn_wg: Tk%8, D%8 required");
  auto out = at::empty({B, Tk, HD}, q.options());
  mxa::attn_wg_bf16(bfp(p), bfp(q), bfp_mut(out), (int)B, (int)H,
                    (int)Tq, (int)Tk, (int)D, zp(p), cur_stream());
  return out;
}

at::Tensor dwconv_fwd(const at::Tensor& x, const at::Tensor& w,
                      const c10::optional<at::Tensor>& bias,
                      std::vector<long> stride, std::vector<long> pad,
                      std::vector<long> dil) {
  // x [N,H,W,C], w [R,S,C] (depthwise, groups == C)
  TORCH_CHECK(x.dim() == 4 && w.dim() == 3 && x.size(3) == w.size(2));
  TORCH_CHECK(x.size(3) % 8 == 0, "dwconv: C % 8 != 0");
  int Nb = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int R = w.size(0), S = w.size(1);
  int sh = stride[0], sw = stride[1], ph = pad[0], pw = pad[1];
  int dh = dil[0], dw_ = dil[1];
  int P = (H + 2 * ph - (R - 1) * dh - 1) / sh + 1;
  int Q = (W + 2 * pw - (S - 1) * dw_ - 1) / sw + 1;
  auto y = at::empty({Nb, P, Q, C}, x.options());
  mxa::dwconv_fwd_bf16(bfp(x), bfp(w),
                       bias.has_value() ? bias->data_ptr<float>() : nullptr,
                       bfp_mut(y), Nb, H, W, C, P, Q, R, S, sh, sw, ph, pw,
                       dh, dw_, cur_stream());
  return y;
}

at::Tensor dwconv_dgrad(const at::Tensor& dy, const at::Tensor& w,
                        std::vector<long> xshape, std::vector<long> stride,
                        std::vector<long> pad) {
  int Nb = xshape[0], H = xshape[1], W = xshape[2], C = xshape[3];
  int P = dy.size(1), Q = dy.size(2);
  int R = w.size(0), S = w.size(1);
  auto dx = at::empty({Nb, H, W, C}, dy.options());
  mxa::dwconv_dgrad_bf16(bfp(dy), bfp(w), bfp_mut(dx), Nb, H, W, C, P, Q,
                         R, S, stride[0], stride[1], pad[0], pad[1],
                         cur_stream());
  return dx;
}

at::Tensor dwconv_wgrad(const at::Tensor& x, const at::Tensor& dy,
                        std::vector<long> rs, std::vector<long> stride,
                        std::vector<long> pad) {
  int Nb = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int P = dy.size(1), Q = dy.size(2);
  int R = rs[0], S = rs[1];
  auto dwf = at::empty({R, S, C}, x.options().dtype(at::kFloat));
  mxa::dwconv_wgrad_bf16(bfp(x), bfp(dy), dwf.data_ptr<float>(), Nb, H, W,
                         C, P, Q, R, S, stride[0], stride[1], pad[0],
                         pad[1], cur_stream());
  return dwf;
}

at::Tensor im2col_pad(const at::Tensor& x, std::vector<long> rs,
                      std::vector<long> stride, std::vector<long> pad,
                      std::vector<long> dil, long kpadded) {
  // stem conv operand: [N*P*Q, kpadded] padded im2col (k order (r,s,c))
  TORCH_CHECK(x.dim() == 4 && x.is_contiguous());
  int Nb = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int R = rs[0], S = rs[1], sh = stride[0], sw = stride[1];
  int ph = pad[0], pw = pad[1], dh = dil[0], dw = dil[1];
  int P = (H + 2 * ph - (R - 1) * dh - 1) / sh + 1;
  int Q = (W + 2 * pw - (S - 1) * dw - 1) / sw + 1;
  int kdim = R * S * C;
  TORCH_CHECK(kpadded % 64 == 0 && kpadded >= kdim);
  auto out = at::empty({(long)Nb * P * Q, kpadded}, x.options());
  mxa::im2col_stem_bf16(bfp(x), bfp_mut(out), Nb, H, W, C, P, Q, R, S, sh,
                        sw, ph, pw, dh, dw, kdim, (int)kpadded,
                        cur_stream());
  return out;
}

at::Tensor colsum(const at::Tensor& x) {
  // bias gradient: sum over all leading dims of a [..., N] bf16 tensor,
  // f32 result (replaces .float().sum(0) copy+reduce pair).
  TORCH_CHECK(x.is_contiguous(), "colsum: contiguous input required");
  int N = (int)x.size(-1);
  long m = x.numel() / N;
  auto out = at::empty({N}, x.options().dtype(at::kFloat));
  mxa::colsum_bf16(bfp(x), out.data_ptr<float>(), m, N, cur_stream());
  return out;
}

std::vector<at::Tensor> dropout_fwd(const at::Tensor& x, double p,
                                    long seed) {
  long n = x.numel();
  TORCH_CHECK(n % 8 == 0, "dropout: numel must be %8");
  auto y = at::empty_like(x);
  auto mask = at::empty({n}, x.options().dtype(at::kByte));
  auto& sw = rng_seed_for(x);
  mxa::dropout_fwd_bf16(bfp(x), bfp_mut(y), mask.data_ptr<unsigned char>(),
                        n,
                        (const unsigned long long*)sw.data_ptr<long>(),
                        (unsigned long long)seed, (float)p, cur_stream());
  return {y, mask};
}

void rng_step(long device) {
  auto opts = at::TensorOptions().device(at::kCUDA, (int)device)
                  .dtype(at::kLong);
  auto probe = at::empty({0}, opts);
  auto& sw = rng_seed_for(probe);
  mxa::rng_bump_launch((unsigned long long*)sw.data_ptr<long>(),
                       cur_stream());
}

at::Tensor dropout_bwd(const at::Tensor& dy, const at::Tensor& mask,
                       double p) {
  long n = dy.numel();
  auto dx = at::empty_like(dy);
  mxa::dropout_bwd_bf16(bfp(dy), mask.data_ptr<unsigned char>(),
                        bfp_mut(dx), n, (float)p, cur_stream());
  return dx;
}

at::Tensor embedding_fwd(const at::Tensor& idx, const at::Tensor& W) {
  auto idx_l = idx.to(at::kLong).contiguous();
  long ntok = idx_l.numel();
  int D = W.size(1);
  TORCH_CHECK(D % 8 == 0, "embedding: dim must be %8");
  auto sizes = idx.sizes().vec();
  sizes.push_back(D);
  auto out = at::empty(sizes, W.options());
  mxa::embedding_fwd_bf16(idx_l.data_ptr<long>(), bfp(W), bfp_mut(out),
                          ntok, D, W.size(0), cur_stream());
  return out;
}

at::Tensor embedding_bwd(const at::Tensor& idx, const at::Tensor& dy,
                         long vocab) {
  auto idx_l = idx.to(at::kLong).contiguous();
  long ntok = idx_l.numel();
  int D = dy.size(-1);
  auto dW = at::empty({vocab, D}, dy.options().dtype(at::kFloat));
  mxa::embedding_bwd_bf16(idx_l.data_ptr<long>(), bfp(dy),
                          dW.data_ptr<float>(), ntok, D, vocab,
                          cur_stream());
  return dW;
}

at::Tensor act_fwd(long op, const at::Tensor& x) {
  long n = x.numel();
  TORCH_CHECK(n % 8 == 0, "act: numel must be %8");
  auto y = at::empty_like(x);
  mxa::act_fwd_bf16((int)op, bfp(x), bfp_mut(y), n, cur_stream());
  return y;
}

at::Tensor act_bwd(long op, const at::Tensor& x, const at::Tensor& dy) {
  long n = x.numel();
  auto dx = at::empty_like(x);
  mxa::act_bwd_bf16((int)op, bfp(x), bfp(dy), bfp_mut(dx), n, cur_stream());
  return dx;
}

// ---------------------------------------------------------------------------
// multi-tensor optimizer
// ---------------------------------------------------------------------------

// Packs descriptors; returns (blob_tensor_on_device, nchunks).
std::vector<at::Tensor> build_opt_descs(std::vector<at::Tensor> ws,
                                        std::vector<at::Tensor> gs,
                                        std::vector<at::Tensor> ms,
                                        std::vector<at::Tensor> vs,
                                        std::vector<at::Tensor> w32s) {
  int n = ws.size();
  std::vector<mxa::TensorDesc> descs(n);
  long chunks = 0;
  for (int i = 0; i < n; ++i) {
    descs[i].w = ws[i].data_ptr();
    descs[i].g = gs[i].data_ptr();
    descs[i].m = ms.empty() || !ms[i].defined() ? nullptr
                                                : ms[i].data_ptr<float>();
    descs[i].v = vs.empty() || !vs[i].defined() ? nullptr
                                                : vs[i].data_ptr<float>();
    descs[i].w32 = w32s.empty() || !w32s[i].defined()
                       ? nullptr
                       : w32s[i].data_ptr<float>();
    descs[i].n = ws[i].numel();
    descs[i].chunk0 = chunks;
    chunks += (descs[i].n + 65535) / 65536;
  }
  auto blob = at::from_blob(descs.data(),
                            {(long)(n * sizeof(mxa::TensorDesc))},
                            at::TensorOptions().dtype(at::kByte))
                  .clone()
                  .to(ws[0].device());
  auto meta = at::tensor({(long)n, chunks}, at::kLong);
  return {blob, meta};
}

void multi_opt_run(const at::Tensor& blob, long ntens, long nchunks,
                   long mode, bool wbf16, double lr, double momentum,
                   double beta1, double beta2, double eps, double wd,
                   double rescale, double clip) {
  mxa::multi_opt(
      reinterpret_cast<const mxa::TensorDesc*>(blob.data_ptr()), (int)ntens,
      nchunks, (int)mode, wbf16, (float)lr, (float)momentum, (float)beta1,
      (float)beta2, (float)eps, (float)wd, (float)rescale, (float)clip,
      cur_stream());
}

at::Tensor multi_sumsq_run(const at::Tensor& blob, long ntens, long nchunks,
                           bool bf16in) {
  auto out = at::zeros({ntens}, blob.options().dtype(at::kFloat));
  mxa::multi_sumsq(
      reinterpret_cast<const mxa::TensorDesc*>(blob.data_ptr()), (int)ntens,
      nchunks, bf16in, out.data_ptr<float>(), cur_stream());
  return out;
}

// ---------------------------------------------------------------------------
// engine: raw HIP stream management (wrapped by torch ExternalStream)
// ---------------------------------------------------------------------------

// hipEvent pool: WaitForVar/cross-stream ordering primitives (reference
// engine.h VarHandle semantics collapse to event waits under stream order)
// ---- baseline JPEG decoder (jpeg.hip) ----
namespace mxa_jpeg {
void decode(const unsigned char*, size_t, std::vector<unsigned char>&,
            int*, int*);
}

at::Tensor imdecode_jpeg(py::bytes data) {
  std::string buf = data;  // copy; decoding reads it on the CPU
  std::vector<unsigned char> rgb;
  int w = 0, h = 0;
  mxa_jpeg::decode((const unsigned char*)buf.data(), buf.size(), rgb, &w,
                   &h);
  auto out = at::empty({(long)h, (long)w, 3},
                       at::TensorOptions().dtype(at::kByte));
  std::memcpy(out.data_ptr<unsigned char>(), rgb.data(), rgb.size());
  return out;
}

// ---- dependency engine (engine.hip: var-queue scheduler) ----
namespace mxa_engine {
uint64_t engine_new_var();
void engine_delete_var(uint64_t);
uint64_t engine_var_version(uint64_t);
void engine_push(std::function<void()>, const std::vector<uint64_t>&,
                 const std::vector<uint64_t>&, int);
void engine_wait_for_var(uint64_t);
void engine_wait_all();
const char* engine_type();
hipStream_t engine_current_stream();
}  // namespace mxa_engine

static void dep_push_py(py::function fn, std::vector<uint64_t> reads,
                        std::vector<uint64_t> writes, int device) {
  // py::function must be destroyed under the GIL; the payload runs on an
  // engine worker thread (GIL acquired around the call)
  std::shared_ptr<py::function> fp(
      new py::function(std::move(fn)), [](py::function* p) {
        py::gil_scoped_acquire g;
        delete p;
      });
  mxa_engine::engine_push(
      [fp]() {
        py::gil_scoped_acquire g;
        (*fp)();
      },
      reads, writes, device);
}

// native payload: y += sum(xs) elementwise (KVStore 'device' reduce) —
// no GIL on the hot path; runs on the engine worker's stream
static void dep_push_sum_py(at::Tensor dst, std::vector<at::Tensor> srcs,
                            std::vector<uint64_t> reads,
                            std::vector<uint64_t> writes, int device) {
  auto payload = [dst, srcs]() {
    c10::InferenceMode guard;
    for (auto& s : srcs) dst.add_(s);
  };
  mxa_engine::engine_push(payload, reads, writes, device);
}

// native payload: in-place allreduce of a tensor list (KVStore local /
// device reduce): acc into t0, optional average, broadcast back.  Runs on
// an engine worker thread with no GIL; cross-device copies go through
// torch's peer copy (xGMI P2P).
static void dep_push_allreduce_py(std::vector<at::Tensor> ts, bool average,
                                  std::vector<uint64_t> reads,
                                  std::vector<uint64_t> writes, int device) {
  auto payload = [ts, average]() {
    c10::InferenceMode guard;
    auto& t0 = const_cast<at::Tensor&>(ts[0]);
    for (size_t i = 1; i < ts.size(); ++i)
      t0.add_(ts[i].to(t0.device(), /*non_blocking=*/true));
    if (average) t0.div_((double)ts.size());
    for (size_t i = 1; i < ts.size(); ++i)
      const_cast<at::Tensor&>(ts[i]).copy_(t0, /*non_blocking=*/true);
  };
  mxa_engine::engine_push(payload, reads, writes, device);
}

long engine_new_event() {
  hipEvent_t e;
  TORCH_CHECK(hipEventCreateWithFlags(&e, hipEventDisableTiming) ==
              hipSuccess);
  return (long)(uintptr_t)e;
}

void engine_record_event(long ev, long stream) {
  TORCH_CHECK(hipEventRecord((hipEvent_t)(uintptr_t)ev,
                             (hipStream_t)(uintptr_t)stream) == hipSuccess);
}

void engine_stream_wait_event(long stream, long ev) {
  TORCH_CHECK(hipStreamWaitEvent((hipStream_t)(uintptr_t)stream,
                                 (hipEvent_t)(uintptr_t)ev, 0) ==
              hipSuccess);
}

void engine_sync_event(long ev) {
  TORCH_CHECK(hipEventSynchronize((hipEvent_t)(uintptr_t)ev) == hipSuccess);
}

void engine_destroy_event(long ev) {
  (void)hipEventDestroy((hipEvent_t)(uintptr_t)ev);
}

long engine_new_stream(long device) {
  hipStream_t s;
  int prev;
  (void)hipGetDevice(&prev);
  (void)hipSetDevice((int)device);
  auto err = hipStreamCreateWithFlags(&s, hipStreamNonBlocking);
  (void)hipSetDevice(prev);
  TORCH_CHECK(err == hipSuccess, "hipStreamCreate failed");
  return (long)(uintptr_t)s;
}

// ---------------------------------------------------------------------------
// C predict API (reference src/c_api/c_predict_api.cc).  extern "C"
// symbols exported from this .so; they drive anand_mxnet_amd.predict
// through the embedded interpreter.  Thread-local error string via
// MXAGetLastError (reference MXGetLastError contract).
// ---------------------------------------------------------------------------
// ---- raw hipGraph step capture (ROADMAP #7: torch's CUDAGraph wrapper
// segfaults on this torch/ROCm combo; this is the direct HIP path).
// Capture must run on a NON-default stream (python wraps the step in
// torch.cuda.stream(side_stream)); replay launches the instantiated exec
// on the current stream.  Allocator stability contract: capture in steady
// state so every tensor the captured kernels touch stays alive/stable.
static std::unordered_map<long, hipGraphExec_t> g_graph_execs;
static long g_graph_next = 1;

long graph_begin() {
  hipStream_t s = cur_stream();
  // relaxed: tolerate event queries from the allocator/autograd engine
  hipError_t e = hipStreamBeginCapture(s, hipStreamCaptureModeRelaxed);
  TORCH_CHECK(e == hipSuccess, "hipStreamBeginCapture: ",
              hipGetErrorString(e));
  return 0;
}

void graph_abort() {
  // recover a stream stuck in capture after a failed step
  hipStream_t s = cur_stream();
  hipStreamCaptureStatus st;
  if (hipStreamIsCapturing(s, &st) == hipSuccess &&
      st != hipStreamCaptureStatusNone) {
    hipGraph_t g = nullptr;
    hipStreamEndCapture(s, &g);
    if (g) hipGraphDestroy(g);
  }
  (void)hipGetLastError();  // clear sticky error
}

long graph_end() {
  hipStream_t s = cur_stream();
  hipGraph_t graph = nullptr;
  hipError_t e = hipStreamEndCapture(s, &graph);
  TORCH_CHECK(e == hipSuccess, "hipStreamEndCapture: ",
              hipGetErrorString(e));
  hipGraphExec_t exec = nullptr;
  e = hipGraphInstantiate(&exec, graph, nullptr, nullptr, 0);
  hipGraphDestroy(graph);
  TORCH_CHECK(e == hipSuccess, "hipGraphInstantiate: ",
              hipGetErrorString(e));
  long h = g_graph_next++;
  g_graph_execs[h] = exec;
  return h;
}

void graph_launch(long h) {
  auto it = g_graph_execs.find(h);
  TORCH_CHECK(it != g_graph_execs.end(), "unknown graph handle");
  hipError_t e = hipGraphLaunch(it->second, cur_stream());
  TORCH_CHECK(e == hipSuccess, "hipGraphLaunch: ", hipGetErrorString(e));
}

void graph_free(long h) {
  auto it = g_graph_execs.find(h);
  if (it != g_graph_execs.end()) {
    hipGraphExecDestroy(it->second);
    g_graph_execs.erase(it);
  }
}

static thread_local std::string mxa_last_error;

extern "C" {

const char* MXAGetLastError() { return mxa_last_error.c_str(); }

int MXAPredCreate(const char* symbol_json, const char* params_file,
                  int dev_type, int dev_id, unsigned num_inputs,
                  const char** input_names, void** out) {
  py::gil_scoped_acquire g;
  try {
    py::module m = py::module::import("anand_mxnet_amd.predict");
    py::list names;
    for (unsigned i = 0; i < num_inputs; ++i)
      names.append(py::str(input_names[i]));
    py::object p = m.attr("CPredictor")(py::str(symbol_json),
                                        py::str(params_file), dev_type,
                                        dev_id, names);
    *out = new py::object(std::move(p));
    return 0;
  } catch (const std::exception& e) {
    mxa_last_error = e.what();
    return -1;
  }
}

int MXAPredSetInput(void* h, const char* name, const float* data,
                    const long* shape, unsigned ndim) {
  py::gil_scoped_acquire g;
  try {
    auto& p = *(py::object*)h;
    std::vector<long> dims(shape, shape + ndim);
    long n = 1;
    for (long d : dims) n *= d;
    py::array_t<float> arr(dims);
    std::memcpy(arr.mutable_data(), data, sizeof(float) * n);
    p.attr("set_input")(py::str(name), arr);
    return 0;
  } catch (const std::exception& e) {
    mxa_last_error = e.what();
    return -1;
  }
}

int MXAPredForward(void* h) {
  py::gil_scoped_acquire g;
  try {
    ((py::object*)h)->attr("forward")();
    return 0;
  } catch (const std::exception& e) {
    mxa_last_error = e.what();
    return -1;
  }
}

int MXAPredGetOutputShape(void* h, unsigned idx, long* shape,
                          unsigned* ndim) {
  py::gil_scoped_acquire g;
  try {
    py::list sh = ((py::object*)h)->attr("output_shape")(idx);
    *ndim = (unsigned)py::len(sh);
    for (unsigned i = 0; i < *ndim; ++i) shape[i] = sh[i].cast<long>();
    return 0;
  } catch (const std::exception& e) {
    mxa_last_error = e.what();
    return -1;
  }
}

int MXAPredGetOutput(void* h, unsigned idx, float* out, long size) {
  py::gil_scoped_acquire g;
  try {
    py::array_t<float> a =
        ((py::object*)h)->attr("output_data")(idx).cast<py::array_t<float>>();
    if ((long)a.size() > size) {
      mxa_last_error = "output buffer too small";
      return -1;
    }
    std::memcpy(out, a.data(), sizeof(float) * a.size());
    return 0;
  } catch (const std::exception& e) {
    mxa_last_error = e.what();
    return -1;
  }
}

int MXAPredFree(void* h) {
  py::gil_scoped_acquire g;
  delete (py::object*)h;
  return 0;
}

}  // extern "C"

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gemm_tn", &gemm_tn, "C = A @ B^T (bf16, MFMA)");
  m.def("gemm_tn_stats", &gemm_tn_stats,
        "gemm_tn + fused per-column BN sums: {out[,stats]}");
  m.def("gemm_wg_bias", &gemm_wg_bias);
  m.def("gemm_wg", &gemm_wg,
        "C(bf16) = sum_m A[m,:] outer B[m,:] (transpose: C[No,Mo])",
        py::arg("A"), py::arg("B"), py::arg("transpose") = false);
  m.def("gemm_tn_fp8", &gemm_tn_fp8);
  m.def("cast_to_fp8", &cast_to_fp8);
  m.def("rng_sample", &rng_sample);
  m.def("ew_unary", &ew_unary_rt);
  m.def("ew_unary_grad", &ew_unary_grad_rt);
  m.def("ew_binary", &ew_binary_rt);
  m.def("ew_reduce_last", &ew_reduce_last_rt);
  m.def("mx_cast", &mx_cast, py::arg("x"), py::arg("e5") = false);
  m.def("gemm_i8", &gemm_i8, py::arg("A8"), py::arg("B8"),
        py::arg("inv_scale") = py::none(), py::arg("bias") = py::none());
  m.def("mx_cast_t", &mx_cast_t, py::arg("x"), py::arg("e5") = false);
  m.def("gemm_mx", &gemm_mx, py::arg("A8"), py::arg("Asc"), py::arg("B8"),
        py::arg("Bsc"), py::arg("bias") = py::none(),
        py::arg("relu") = false, py::arg("afmt") = 0, py::arg("bfmt") = 0);
  m.def("conv2d_fwd", &conv2d_fwd);
  m.def("conv2d_fwd_stats", &conv2d_fwd_stats,
        "conv2d_fwd + fused per-channel BN sums: {y[,stats]}");
  m.def("conv2d_dgrad", &conv2d_dgrad);
  m.def("conv2d_wgrad", &conv2d_wgrad);
  m.def("bn_fwd_train", &bn_fwd_train, py::arg("x"), py::arg("gamma"),
        py::arg("beta"), py::arg("running_mean"), py::arg("running_var"),
        py::arg("eps"), py::arg("momentum"), py::arg("relu"),
        py::arg("resid") = py::none(), py::arg("psums") = py::none());
  m.def("bn_fwd_infer", &bn_fwd_infer, py::arg("x"), py::arg("gamma"),
        py::arg("beta"), py::arg("rmean"), py::arg("rvar"),
        py::arg("eps"), py::arg("relu"), py::arg("resid") = py::none());
  m.def("bn_bwd", &bn_bwd);
  m.def("maxpool", &maxpool);
  m.def("maxpool_bwd", &maxpool_bwd);
  m.def("avgpool", &avgpool);
  m.def("global_avgpool", &global_avgpool);
  m.def("global_avgpool_bwd", &global_avgpool_bwd);
  m.def("softmax_fwd", &softmax_fwd, py::arg("x"), py::arg("log_"),
        py::arg("alpha") = 1.0);
  m.def("softmax_masked_fwd", &softmax_masked_fwd, py::arg("x"),
        py::arg("vlen"), py::arg("alpha") = 1.0);
  m.def("softmax_bwd", &softmax_bwd, py::arg("y"), py::arg("dy"),
        py::arg("log_"), py::arg("alpha") = 1.0);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("colsum", &colsum);
  m.def("im2col_pad", &im2col_pad);
  m.def("dwconv_fwd", &dwconv_fwd);
  m.def("dwconv_dgrad", &dwconv_dgrad);
  m.def("dwconv_wgrad", &dwconv_wgrad);
  m.def("gemm_nn", &gemm_nn);
  m.def("attn_scores", &attn_scores);
  m.def("attn_ctx", &attn_ctx);
  m.def("attn_wg", &attn_wg);
  m.def("flash_fwd", &flash_fwd);
  m.def("flash_bwd", &flash_bwd);
  m.def("dropout_fwd", &dropout_fwd);
  m.def("rng_step", &rng_step);
  m.def("softmax_dropout_fwd", &softmax_dropout_fwd);
  m.def("softmax_dropout_bwd", &softmax_dropout_bwd);
  m.def("dropout_bwd", &dropout_bwd);
  m.def("embedding_fwd", &embedding_fwd);
  m.def("embedding_bwd", &embedding_bwd);
  m.def("act_fwd", &act_fwd);
  m.def("act_bwd", &act_bwd);
  m.def("build_opt_descs", &build_opt_descs);
  m.def("multi_opt_run", &multi_opt_run);
  m.def("multi_sumsq_run", &multi_sumsq_run);
  m.def("engine_new_stream", &engine_new_stream);
  m.def("dep_new_var", &mxa_engine::engine_new_var);
  m.def("dep_delete_var", &mxa_engine::engine_delete_var);
  m.def("dep_var_version", &mxa_engine::engine_var_version);
  m.def("dep_push", &dep_push_py, py::arg("fn"), py::arg("reads"),
        py::arg("writes"), py::arg("device") = -1);
  m.def("dep_push_sum", &dep_push_sum_py);
  m.def("dep_push_allreduce", &dep_push_allreduce_py, py::arg("tensors"),
        py::arg("average"), py::arg("reads"), py::arg("writes"),
        py::arg("device") = -1);
  m.def("dep_wait_for_var", &mxa_engine::engine_wait_for_var,
        py::call_guard<py::gil_scoped_release>());
  m.def("dep_wait_all", &mxa_engine::engine_wait_all,
        py::call_guard<py::gil_scoped_release>());
  m.def("imdecode_jpeg", &imdecode_jpeg, "baseline JPEG -> HWC uint8");
  m.def("graph_begin", &graph_begin);
  m.def("graph_abort", &graph_abort);
  m.def("graph_end", &graph_end);
  m.def("graph_launch", &graph_launch);
  m.def("graph_free", &graph_free);
  m.def("dep_engine_type",
        []() { return std::string(mxa_engine::engine_type()); });
  m.def("engine_new_event", &engine_new_event);
  m.def("engine_record_event", &engine_record_event);
  m.def("engine_stream_wait_event", &engine_stream_wait_event);
  m.def("engine_sync_event", &engine_sync_event);
  m.def("engine_destroy_event", &engine_destroy_event);
}
