// DeepConsensus-AMD HIP/CDNA4 kernels (gfx950 / MI355X).
//
// Kernel inventory:
//  * fused_embed_condense — K2+K3: the subread-stack embedding gathers
//    (reference networks.py:436-516) fused with the no-bias condenser GEMM
//    (networks.py:426-434) via precomputed per-row fused tables
//    T'_row[id] = (table_f[id] * sqrt(w) * (id != 0)) @ Wc[row_slice].
//    One pass over the [B, R, L] feature tensor emits [B, L, H] bf16 directly,
//    never materializing the 560-wide concat.
//  * fused_ln_head_qv — K10+K11+K12: final LayerNorm (eps 1e-6, fp32,
//    encoder_stack.py:131-133) + Dense(5) head (networks.py:342-345) +
//    softmax + argmax + Phred QV with linear calibration and cap
//    (quick_inference.py:377-389) fused into one wave-per-position kernel
//    emitting 2 bytes per position (base id + integer QV).
//
// Written CDNA4-first: 64-wide wavefronts, LDS staging for the id tile,
// fp32 accumulation, no CUDA-compat shims.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DC_CHECK(x, msg) TORCH_CHECK(x, msg)

using bf16 = __hip_bfloat16;

// ---------------------------------------------------------------------------
// fused_embed_condense
// ---------------------------------------------------------------------------
// rows:        [B, R, L] float32 (feature values; clipped upstream)
// fused_table: [V_total, H] bf16
// row_offset:  [R] int32  start row of this input row's fused table
// row_shift:   [R] int32  added to the raw value before lookup (ccs_bq: +1)
// row_vocab:   [R] int32  vocab size of this row's table (clamp bound)
// out:         [B, L, H] bf16
//
// Geometry: one workgroup = TPP*TILE_L threads = one (b, l-tile); ids staged
// in LDS; each thread owns NDIM output dims of one position, accumulating in
// fp32 VGPRs with 16-byte bf16x8 table loads.
namespace {

constexpr int TILE_L = 64;   // positions per workgroup
constexpr int TPP = 5;       // threads per position
constexpr int NDIM = 56;     // dims per thread (TPP * NDIM = 280)

__global__ __launch_bounds__(TILE_L * TPP) void fused_embed_condense_kernel(
    const float* __restrict__ rows,
    const bf16* __restrict__ fused_table,
    const int* __restrict__ row_offset,
    const int* __restrict__ row_shift,
    const int* __restrict__ row_vocab,
    bf16* __restrict__ out,
    int B, int R, int L, int H) {
  __shared__ int ids[128][TILE_L];  // R <= 128 supported (max_passes <= 30)

  const int tiles_per_b = (L + TILE_L - 1) / TILE_L;
  const int b = blockIdx.x / tiles_per_b;
  const int l0 = (blockIdx.x % tiles_per_b) * TILE_L;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;  // 5 waves

  // Stage the id tile: wave w loads rows r = w, w+5, ... (64 lanes = 64 l's).
  const float* rows_b = rows + (size_t)b * R * L;
  for (int r = wave; r < R; r += TILE_L * TPP / 64) {
    int l = l0 + lane;
    float v = (l < L) ? rows_b[(size_t)r * L + l] : 0.f;
    int id = (int)v + row_shift[r];
    int vmax = row_vocab[r] - 1;
    id = id < 0 ? 0 : (id > vmax ? vmax : id);
    ids[r][lane] = row_offset[r] + id;
  }
  __syncthreads();

  // Accumulate: thread handles position p = tid / TPP, dim chunk c = tid % TPP.
  const int p = tid / TPP;
  const int c = tid % TPP;
  const int l = l0 + p;
  if (l >= L) return;

  float acc[NDIM];
#pragma unroll
  for (int d = 0; d < NDIM; ++d) acc[d] = 0.f;

  const int dim0 = c * NDIM;
  for (int r = 0; r < R; ++r) {
    const bf16* trow = fused_table + (size_t)ids[r][p] * H + dim0;
    // NDIM = 56 bf16 = 7 x 16-byte loads.
#pragma unroll
    for (int v = 0; v < NDIM / 8; ++v) {
      // 16B vector load of 8 bf16.
      const uint4 raw = *reinterpret_cast<const uint4*>(trow + v * 8);
      const bf16* e = reinterpret_cast<const bf16*>(&raw);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[v * 8 + j] += __bfloat162float(e[j]);
    }
  }

  bf16* orow = out + ((size_t)b * L + l) * H + dim0;
#pragma unroll
  for (int v = 0; v < NDIM / 8; ++v) {
    uint4 raw;
    bf16* e = reinterpret_cast<bf16*>(&raw);
#pragma unroll
    for (int j = 0; j < 8; ++j) e[j] = __float2bfloat16(acc[v * 8 + j]);
    *reinterpret_cast<uint4*>(orow + v * 8) = raw;
  }
}

}  // namespace

at::Tensor fused_embed_condense(
    at::Tensor rows, at::Tensor fused_table, at::Tensor row_offset,
    at::Tensor row_shift, at::Tensor row_vocab) {
  DC_CHECK(rows.is_cuda() && rows.dtype() == at::kFloat,
           "rows must be float32 on device");
  DC_CHECK(fused_table.dtype() == at::kBFloat16, "fused_table must be bf16");
  DC_CHECK(rows.is_contiguous() && fused_table.is_contiguous(),
           "contiguous inputs required");
  const int B = rows.size(0), R = rows.size(1), L = rows.size(2);
  const int H = fused_table.size(1);
  DC_CHECK(R <= 128, "R must be <= 128");
  DC_CHECK(H % (TPP * 8) == 0 && H / TPP == NDIM,
           "H must equal 280 for this kernel build");
  auto out = at::empty({B, L, H}, rows.options().dtype(at::kBFloat16));
  const int tiles_per_b = (L + TILE_L - 1) / TILE_L;
  dim3 grid(B * tiles_per_b);
  dim3 block(TILE_L * TPP);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(fused_embed_condense_kernel, grid, block, 0, stream,
                     rows.data_ptr<float>(),
                     reinterpret_cast<bf16*>(fused_table.data_ptr()),
                     row_offset.data_ptr<int>(), row_shift.data_ptr<int>(),
                     row_vocab.data_ptr<int>(),
                     reinterpret_cast<bf16*>(out.data_ptr()), B, R, L, H);
  return out;
}

// ---------------------------------------------------------------------------
// fused_ln_head_qv
// ---------------------------------------------------------------------------
// x:      [N, H] bf16 or fp32 (final encoder output, N = B*L)
// gamma, beta: [H] fp32 (output LayerNorm params)
// w_head: [5, H] fp32, b_head: [5] fp32
// calib:  (threshold, w, b) linear QV calibration; max_qual cap.
// bases_out: [N] uint8 (vocab id), quals_out: [N] uint8
// probs_out: optional [N, 5] fp32 (for eval paths)
//
// One 64-lane wave per position: lanes split the H dims, shuffle-reduce the
// mean/var and the 5 head dots, lane 0 finishes softmax+QV.
namespace {

template <typename T>
__global__ __launch_bounds__(256) void fused_ln_head_qv_kernel(
    const T* __restrict__ x, const float* __restrict__ gamma,
    const float* __restrict__ beta, const float* __restrict__ w_head,
    const float* __restrict__ b_head, uint8_t* __restrict__ bases_out,
    uint8_t* __restrict__ quals_out, float* __restrict__ probs_out,
    int N, int H, float cal_threshold, float cal_w, float cal_b,
    float max_qual) {
  const int wave_in_block = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int n = blockIdx.x * 4 + wave_in_block;
  if (n >= N) return;

  const T* xi = x + (size_t)n * H;

  // Per-lane partial sums over dims lane, lane+64, ...
  float sum = 0.f, sumsq = 0.f;
  float xv[8];  // H <= 512 supported
  int nd = 0;
  for (int d = lane; d < H; d += 64, ++nd) {
    float v;
    if constexpr (std::is_same_v<T, bf16>) v = __bfloat162float(xi[d]);
    else v = xi[d];
    xv[nd] = v;
    sum += v;
    sumsq += v * v;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    sum += __shfl_xor(sum, off, 64);
    sumsq += __shfl_xor(sumsq, off, 64);
  }
  const float mean = sum / H;
  const float var = sumsq / H - mean * mean;
  const float rstd = rsqrtf(var + 1e-6f);

  // Head dots: 5 logits.
  float logit[5];
#pragma unroll
  for (int k = 0; k < 5; ++k) logit[k] = 0.f;
  nd = 0;
  for (int d = lane; d < H; d += 64, ++nd) {
    const float normed = (xv[nd] - mean) * rstd * gamma[d] + beta[d];
#pragma unroll
    for (int k = 0; k < 5; ++k) logit[k] += normed * w_head[k * H + d];
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
#pragma unroll
    for (int k = 0; k < 5; ++k) logit[k] += __shfl_xor(logit[k], off, 64);
  }

  if (lane == 0) {
    float mx = -1e30f;
#pragma unroll
    for (int k = 0; k < 5; ++k) {
      logit[k] += b_head[k];
      mx = fmaxf(mx, logit[k]);
    }
    float denom = 0.f;
    float ex[5];
#pragma unroll
    for (int k = 0; k < 5; ++k) {
      ex[k] = __expf(logit[k] - mx);
      denom += ex[k];
    }
    int arg = 0;
    float pmax = 0.f;
#pragma unroll
    for (int k = 0; k < 5; ++k) {
      const float p = ex[k] / denom;
      if (probs_out != nullptr) probs_out[(size_t)n * 5 + k] = p;
      if (p > pmax) { pmax = p; arg = k; }
    }
    // QV: -10*log10(1-p), linear calibration, cap, banker's round, floor 0
    // (quick_inference.py:377-389).
    float ep = 1.f - pmax;
    float q = (ep <= 0.f) ? 1e9f : -10.f * log10f(ep);
    if (cal_threshold == 0.f || q > cal_threshold) q = q * cal_w + cal_b;
    q = fminf(q, max_qual);
    q = nearbyintf(q);
    q = fmaxf(q, 0.f);
    bases_out[n] = (uint8_t)arg;
    quals_out[n] = (uint8_t)q;
  }
}

}  // namespace

std::vector<at::Tensor> fused_ln_head_qv(
    at::Tensor x, at::Tensor gamma, at::Tensor beta, at::Tensor w_head,
    at::Tensor b_head, double cal_threshold, double cal_w, double cal_b,
    double max_qual, bool want_probs) {
  DC_CHECK(x.is_cuda() && x.dim() >= 2, "x must be a device matrix");
  auto xc = x.contiguous();
  const int H = xc.size(-1);
  const int N = xc.numel() / H;
  DC_CHECK(H <= 512, "H must be <= 512");
  auto opts = xc.options();
  auto bases = at::empty({N}, opts.dtype(at::kByte));
  auto quals = at::empty({N}, opts.dtype(at::kByte));
  at::Tensor probs;
  float* probs_ptr = nullptr;
  if (want_probs) {
    probs = at::empty({N, 5}, opts.dtype(at::kFloat));
    probs_ptr = probs.data_ptr<float>();
  }
  dim3 grid((N + 3) / 4);
  dim3 block(256);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  auto gc = gamma.contiguous(), bc = beta.contiguous();
  auto wc = w_head.contiguous(), bhc = b_head.contiguous();
  if (xc.dtype() == at::kBFloat16) {
    hipLaunchKernelGGL(fused_ln_head_qv_kernel<bf16>, grid, block, 0, stream,
                       reinterpret_cast<bf16*>(xc.data_ptr()),
                       gc.data_ptr<float>(), bc.data_ptr<float>(),
                       wc.data_ptr<float>(), bhc.data_ptr<float>(),
                       bases.data_ptr<uint8_t>(), quals.data_ptr<uint8_t>(),
                       probs_ptr, N, H, (float)cal_threshold, (float)cal_w,
                       (float)cal_b, (float)max_qual);
  } else {
    DC_CHECK(xc.dtype() == at::kFloat, "x must be bf16 or fp32");
    hipLaunchKernelGGL(fused_ln_head_qv_kernel<float>, grid, block, 0, stream,
                       xc.data_ptr<float>(), gc.data_ptr<float>(),
                       bc.data_ptr<float>(), wc.data_ptr<float>(),
                       bhc.data_ptr<float>(), bases.data_ptr<uint8_t>(),
                       quals.data_ptr<uint8_t>(), probs_ptr, N, H,
                       (float)cal_threshold, (float)cal_w, (float)cal_b,
                       (float)max_qual);
  }
  if (want_probs) return {bases, quals, probs};
  return {bases, quals};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_embed_condense", &fused_embed_condense,
        "Fused subread-stack embedding + condenser (K2+K3)");
  m.def("fused_ln_head_qv", &fused_ln_head_qv,
        "Fused final LayerNorm + head + softmax + QV (K10+K11+K12)");
}
