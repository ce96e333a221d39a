// Fused FFN (K9 + ReZero residual K8) for gfx950.
//
// out = x + alpha * (relu(x @ W1^T + b1) @ W2^T + b2)     (ffn_layer.py:69-87
// + encoder_stack.py:88-92), computed tile-resident: the [M, 2048] hidden
// activation never touches HBM (the eager path writes + re-reads ~3.3 GB of
// it per layer at batch 4096, plus a full ReLU pass).
//
// Structure (512 threads = 8 waves; one workgroup = 128 rows):
//  * x tile staged once through LDS, then each wave holds its 18 A-fragments
//    (32 rows x 288 k) in registers; the x LDS region is then REUSED as the
//    weight staging buffer — full-cacheline coalesced staging, never
//    fragment-shaped global reads (cdna_hip_programming.md section 5,
//    "x through LDS in full lines");
//  * 16 chunks of 128 hidden columns: stage W1 slice [128][288] -> barrier
//    -> h_chunk = relu(x @ W1^T + b1) via v_mfma_f32_32x32x16_bf16, written
//    bf16 row-major to the h region -> stage W2 slice [280][128] over the
//    same weight region -> barrier -> out accumulators (5 col-tiles/wave,
//    fp32) consume h_chunk;
//  * epilogue folds b2, the ReZero alpha and the residual (x re-read from
//    global).
// Weights arrive host-padded (W1 [2048, 288], W2 [320, 2048]) so every 16-B
// fragment read is in bounds.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

using bf16 = __hip_bfloat16;

namespace {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

constexpr int BM = 128;        // rows per workgroup
constexpr int K1 = 280;        // model width
constexpr int K1P = 288;       // padded k (W1 second dim)
constexpr int A_STRIDE = 312;  // region-A row stride: 156 dwords % 64 = 28,
                               // gcd(28,64)=4 -> conflict-free b128 groups
constexpr int NC = 128;        // hidden cols per chunk
constexpr int NHID = 2048;
constexpr int NCHUNK = NHID / NC;
constexpr int H_STRIDE = 136;  // h LDS row stride
constexpr int W2_ROWS = 280;   // W2 rows staged (out cols)
constexpr int NOUT_PAD = 320;  // padded W2 first dim (host tensor)

__global__ __launch_bounds__(512, 2) void fused_ffn_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w1,
    const float* __restrict__ b1, const bf16* __restrict__ w2,
    const float* __restrict__ b2, bf16* __restrict__ out,
    int M, float alpha) {
  // Region A (77.8 KB): x image, then per-chunk W1 [128][A_STRIDE] and
  // W2 [280][H_STRIDE] images. Region H (34.8 KB): hidden chunk.
  __shared__ __attribute__((aligned(16))) bf16 smem_a[BM * A_STRIDE];
  __shared__ __attribute__((aligned(16))) bf16 h_lds[BM][H_STRIDE];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;      // 0..7
  const int lane = tid & 63;
  const int c = lane & 31;
  const int hi = lane >> 5;
  const int rg = wave >> 1;       // row group (rows 32*rg..+31)
  const int ch = wave & 1;        // out col half (cols 140*ch..+139)
  const int m0 = blockIdx.x * BM;

  // ---- Stage x tile into region A (zero pad cols/rows), coalesced. ----
  for (int idx = tid; idx < BM * (A_STRIDE / 2); idx += 512) {
    const int r = idx / (A_STRIDE / 2), d2 = idx % (A_STRIDE / 2);
    unsigned v = 0;
    if (m0 + r < M && 2 * d2 + 1 < K1) {
      v = *reinterpret_cast<const unsigned*>(
          x + (size_t)(m0 + r) * K1 + 2 * d2);
    }
    *reinterpret_cast<unsigned*>(&smem_a[r * A_STRIDE + 2 * d2]) = v;
  }
  __syncthreads();

  // ---- Pull this wave's 18 x A-fragments into registers. ----
  bf16x8 af[18];
#pragma unroll
  for (int s = 0; s < 18; ++s) {
    af[s] = *reinterpret_cast<const bf16x8*>(
        &smem_a[(32 * rg + c) * A_STRIDE + 16 * s + 8 * hi]);
  }
  __syncthreads();  // region A free for weight staging

  f32x16 oacc[5] = {};

  for (int chunk = 0; chunk < NCHUNK; ++chunk) {
    const int n0 = chunk * NC;
    // ---- Stage W1 slice [NC rows n0..][K1P] -> region A, coalesced. ----
    for (int g = tid; g < NC * (K1P / 8); g += 512) {
      const int row = g / (K1P / 8), k8 = g % (K1P / 8);
      *reinterpret_cast<uint4*>(&smem_a[row * A_STRIDE + 8 * k8]) =
          *reinterpret_cast<const uint4*>(
              w1 + (size_t)(n0 + row) * K1P + 8 * k8);
    }
    __syncthreads();

    // ---- B1: h_chunk = relu(x @ W1^T + b1). Wave: rows 32*rg..+31,
    // hidden cols 64*ch + {0..63} (two 32-col tiles). ----
#pragma unroll 1
    for (int t = 0; t < 2; ++t) {
      const int colt = 64 * ch + 32 * t;  // within chunk
      f32x16 acc = {};
#pragma unroll
      for (int s = 0; s < 18; ++s) {
        const bf16x8 bfr = *reinterpret_cast<const bf16x8*>(
            &smem_a[(colt + c) * A_STRIDE + 16 * s + 8 * hi]);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af[s], bfr, acc,
                                                      0, 0, 0);
      }
      const float bias = b1[n0 + colt + c];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int row = 32 * rg + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float v = acc[r] + bias;
        v = v > 0.f ? v : 0.f;
        h_lds[row][colt + c] = __float2bfloat16(v);
      }
    }
    __syncthreads();

    // ---- Stage W2 slice [280 rows][NC k] over region A, coalesced. ----
    for (int g = tid; g < W2_ROWS * (NC / 8); g += 512) {
      const int row = g / (NC / 8), k8 = g % (NC / 8);
      *reinterpret_cast<uint4*>(&smem_a[row * H_STRIDE + 8 * k8]) =
          *reinterpret_cast<const uint4*>(
              w2 + (size_t)row * NHID + n0 + 8 * k8);
    }
    __syncthreads();

    // ---- B2: oacc += h_chunk @ W2^T (this wave's 140 out cols). ----
#pragma unroll 1
    for (int s = 0; s < NC / 16; ++s) {
      const bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &h_lds[32 * rg + c][16 * s + 8 * hi]);
#pragma unroll
      for (int ct = 0; ct < 5; ++ct) {
        const int ocol = min(140 * ch + 32 * ct + c, W2_ROWS - 1);
        const bf16x8 bfr = *reinterpret_cast<const bf16x8*>(
            &smem_a[ocol * H_STRIDE + 16 * s + 8 * hi]);
        oacc[ct] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, bfr,
                                                           oacc[ct], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- Epilogue: out = x + alpha * (oacc + b2); x re-read from global.
  // Fully unrolled: a runtime-indexed oacc[ct] would force the accumulators
  // to scratch for the whole kernel (cdna_hip_programming.md rule 20). ----
#pragma unroll
  for (int ct = 0; ct < 5; ++ct) {
    const int col = 140 * ch + 32 * ct + c;
    if (col >= K1) continue;
    const float bias = b2[col];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = 32 * rg + (r & 3) + 8 * (r >> 2) + 4 * hi;
      if (m0 + row < M) {
        const size_t off = (size_t)(m0 + row) * K1 + col;
        const float resid = __bfloat162float(x[off]);
        out[off] = __float2bfloat16(resid + alpha * (oacc[ct][r] + bias));
      }
    }
  }
}

}  // namespace

at::Tensor fused_ffn(at::Tensor x, at::Tensor w1, at::Tensor b1,
                     at::Tensor w2, at::Tensor b2, double alpha) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16,
              "x must be bf16 on device");
  auto xc = x.contiguous();
  const int K = xc.size(-1);
  const int M = xc.numel() / K;
  TORCH_CHECK(K == K1, "fused_ffn requires width 280");
  TORCH_CHECK(w1.size(0) == NHID && w1.size(1) == K1P,
              "w1 must be padded [2048, 288]");
  TORCH_CHECK(w2.size(0) == NOUT_PAD && w2.size(1) == NHID,
              "w2 must be padded [320, 2048]");
  TORCH_CHECK(b1.numel() == NHID && b2.numel() == NOUT_PAD,
              "bias shapes");
  auto out = at::empty_like(xc);
  dim3 grid((M + BM - 1) / BM);
  dim3 block(512);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(fused_ffn_kernel, grid, block, 0, stream,
                     reinterpret_cast<bf16*>(xc.data_ptr()),
                     reinterpret_cast<bf16*>(w1.data_ptr()),
                     b1.data_ptr<float>(),
                     reinterpret_cast<bf16*>(w2.data_ptr()),
                     b2.data_ptr<float>(),
                     reinterpret_cast<bf16*>(out.data_ptr()),
                     M, (float)alpha);
  return out;
}
