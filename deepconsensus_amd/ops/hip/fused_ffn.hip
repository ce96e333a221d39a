// Fused FFN (K9 + ReZero residual K8) for gfx950.
//
// out = x + alpha * (relu(x @ W1^T + b1) @ W2^T + b2)     (ffn_layer.py:69-87
// + encoder_stack.py:88-92), computed tile-resident: the [M, 2048] hidden
// activation never touches HBM (the eager path writes + re-reads ~3.3 GB of
// it per layer at batch 4096, plus a full ReLU pass).
//
// Structure (512 threads = 8 waves; one workgroup = 128 rows; 32 chunks of
// 64 hidden columns):
//  * x staged once through LDS, then each wave's 18 A-fragments (32 rows x
//    288 k) live in registers; the x image region is re-carved into W1/W2
//    staging buffers;
//  * weight staging is the T14 async-STAGE split
//    (cdna_hip_programming.md G15): each phase ISSUES the next slice's
//    global loads into ~5 uint4 registers before its MFMA cluster and
//    ds_writes them after it, so HBM/L2 latency hides under the matrix
//    work; all LDS images are padded to conflict-free b128 strides;
//  * B1: h_chunk = relu(x @ W1^T + b1) -> bf16 row-major LDS;
//    B2: fp32 out accumulators (5 col-tiles/wave) consume h_chunk vs W2;
//  * epilogue folds b2, the ReZero alpha and the residual.
// Weights arrive host-padded (W1 [2048, 288], W2 [320, 2048]) so every 16-B
// read is in bounds.

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
constexpr int NC = 64;         // hidden cols per chunk
constexpr int NHID = 2048;
constexpr int NCHUNK = NHID / NC;
constexpr int W1_STRIDE = 296;  // 148 dw % 64 = 20 -> conflict-free groups
constexpr int W2_STRIDE = 72;   // 36 dw % 64 = 36 -> conflict-free groups
constexpr int H_STRIDE = 72;
constexpr int W2_ROWS = 280;
constexpr int NOUT_PAD = 320;

constexpr int W1_GRAN = NC * (K1P / 8);       // 16B granules per W1 slice
constexpr int W2_GRAN = W2_ROWS * (NC / 8);   // granules per W2 slice
constexpr int G_PER_T = 5;                    // ceil(granules / 512)

__global__ __launch_bounds__(512, 2) void fused_ffn_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w1,
    const float* __restrict__ b1, const bf16* __restrict__ w2,
    const float* __restrict__ b2, bf16* __restrict__ out,
    int M, float alpha) {
  // W1 buf 37.9 KB + W2 buf 40.3 KB (the x image overlays both at start),
  // h 18.4 KB.
  __shared__ __attribute__((aligned(16))) bf16 smem_w[64 * W1_STRIDE
                                                      + NOUT_PAD * W2_STRIDE];
  __shared__ __attribute__((aligned(16))) bf16 h_lds[BM][H_STRIDE];
  bf16* w1_lds = smem_w;                       // [64][W1_STRIDE]
  bf16* w2_lds = smem_w + 64 * W1_STRIDE;      // [<=320][W2_STRIDE]

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int c = lane & 31;
  const int hi = lane >> 5;
  const int rg = wave >> 1;       // row group (rows 32*rg..+31)
  const int ch = wave & 1;        // col half
  const int m0 = blockIdx.x * BM;

  // ---- Stage x image over smem_w ([128][296] rows), pull A-frags. ----
  // 16-B granules (37/row; row stride 592 B = 37 x 16 B): keeps the x loads
  // in flight instead of serializing ~37 scalar-load latencies.
  for (int idx = tid; idx < BM * 37; idx += 512) {
    const int r = idx / 37, q4 = idx % 37;
    uint4 v = {};
    if (m0 + r < M && 8 * q4 + 8 <= K1) {
      v = *reinterpret_cast<const uint4*>(
          x + (size_t)(m0 + r) * K1 + 8 * q4);
    }
    *reinterpret_cast<uint4*>(&smem_w[r * W1_STRIDE + 8 * q4]) = v;
  }
  __syncthreads();
  bf16x8 af[18];
#pragma unroll
  for (int s = 0; s < 18; ++s) {
    af[s] = *reinterpret_cast<const bf16x8*>(
        &smem_w[(32 * rg + c) * W1_STRIDE + 16 * s + 8 * hi]);
  }
  __syncthreads();

  // ---- T14 staging: issue loads to regs early, ds_write after MFMAs.
  // One shared register set: the W1 and W2 in-flight windows never overlap
  // (issue_w2..write_w2 inside B1; issue_w1..write_w1 inside B2).
  // NAMED scalars, not an array: AMDGPUPromoteAlloca silently moves a
  // per-thread uint4[5] into LDS (5*16B*512 threads = 40 KB!), turning the
  // register pipeline into an LDS round-trip. ----
  uint4 sr0, sr1, sr2, sr3, sr4;
  uint4* stage_reg_ptrs[G_PER_T] = {&sr0, &sr1, &sr2, &sr3, &sr4};
  (void)stage_reg_ptrs;
#define STAGE_REG(i) (i == 0 ? sr0 : i == 1 ? sr1 : i == 2 ? sr2 \
                      : i == 3 ? sr3 : sr4)

  auto issue_w1 = [&](int chunk) {
#pragma unroll
    for (int i = 0; i < G_PER_T; ++i) {
      const int g = min(tid + i * 512, W1_GRAN - 1);
      const int row = g / (K1P / 8), k8 = g % (K1P / 8);
      const uint4 v = *reinterpret_cast<const uint4*>(
          w1 + (size_t)(chunk * NC + row) * K1P + 8 * k8);
      if (i == 0) sr0 = v; else if (i == 1) sr1 = v; else if (i == 2) sr2 = v;
      else if (i == 3) sr3 = v; else sr4 = v;
    }
  };
  auto write_w1 = [&]() {
#pragma unroll
    for (int i = 0; i < G_PER_T; ++i) {
      const int g = min(tid + i * 512, W1_GRAN - 1);
      const int row = g / (K1P / 8), k8 = g % (K1P / 8);
      *reinterpret_cast<uint4*>(&w1_lds[row * W1_STRIDE + 8 * k8]) =
          (i == 0 ? sr0 : i == 1 ? sr1 : i == 2 ? sr2 : i == 3 ? sr3 : sr4);
    }
  };
  auto issue_w2 = [&](int chunk) {
#pragma unroll
    for (int i = 0; i < G_PER_T; ++i) {
      const int g = min(tid + i * 512, W2_GRAN - 1);
      const int row = g / (NC / 8), k8 = g % (NC / 8);
      const uint4 v = *reinterpret_cast<const uint4*>(
          w2 + (size_t)row * NHID + chunk * NC + 8 * k8);
      if (i == 0) sr0 = v; else if (i == 1) sr1 = v; else if (i == 2) sr2 = v;
      else if (i == 3) sr3 = v; else sr4 = v;
    }
  };
  auto write_w2 = [&]() {
#pragma unroll
    for (int i = 0; i < G_PER_T; ++i) {
      const int g = min(tid + i * 512, W2_GRAN - 1);
      const int row = g / (NC / 8), k8 = g % (NC / 8);
      *reinterpret_cast<uint4*>(&w2_lds[row * W2_STRIDE + 8 * k8]) =
          (i == 0 ? sr0 : i == 1 ? sr1 : i == 2 ? sr2 : i == 3 ? sr3 : sr4);
    }
  };

  // Prologue: W1_0 staged synchronously.
  issue_w1(0);
  write_w1();
  __syncthreads();

  f32x16 oacc[5] = {};

  for (int chunk = 0; chunk < NCHUNK; ++chunk) {
    // ---- B1 phase: issue W2_c early; MFMAs; write h + W2_c. ----
    issue_w2(chunk);
    const int colt = 32 * ch;             // wave's hidden col tile
    const int hcol = chunk * NC + colt + c;
    f32x16 acc = {};
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int s = 0; s < 18; ++s) {
      const bf16x8 bfr = *reinterpret_cast<const bf16x8*>(
          &w1_lds[(colt + c) * W1_STRIDE + 16 * s + 8 * hi]);
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af[s], bfr, acc,
                                                    0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    const float bias = b1[hcol];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = 32 * rg + (r & 3) + 8 * (r >> 2) + 4 * hi;
      float v = acc[r] + bias;
      v = v > 0.f ? v : 0.f;
      h_lds[row][colt + c] = __float2bfloat16(v);
    }
    write_w2();
    __syncthreads();

    // ---- B2 phase: issue W1_{c+1} early; MFMAs; write W1_{c+1}. ----
    if (chunk + 1 < NCHUNK) issue_w1(chunk + 1);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int s = 0; s < NC / 16; ++s) {
      const bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &h_lds[32 * rg + c][16 * s + 8 * hi]);
#pragma unroll
      for (int ct = 0; ct < 5; ++ct) {
        const int ocol = min(140 * ch + 32 * ct + c, W2_ROWS - 1);
        const bf16x8 bfr = *reinterpret_cast<const bf16x8*>(
            &w2_lds[ocol * W2_STRIDE + 16 * s + 8 * hi]);
        oacc[ct] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a, bfr, oacc[ct], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    if (chunk + 1 < NCHUNK) write_w1();
    __syncthreads();
  }

  // ---- Epilogue: out = x + alpha * (oacc + b2). Fully unrolled (a
  // runtime-indexed oacc[ct] would force scratch, rule 20). ----
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
