// Fused linear projection for gfx950: out = act(x @ W^T + b)[*alpha + x].
//
// The K5 QKV projection and K7 output projection (attention_layer.py:78-107)
// as one tiled MFMA kernel reusing the fused-FFN structure: 512 threads =
// 8 waves per 128-row tile, K fixed at the model width 280 (padded 288),
// N processed in 64-column chunks with glds weight streaming (120 VGPRs
// -> 4 waves/SIMD occupancy), outputs staged through LDS for coalesced
// 16-B global writes. Optional fused bias, ReLU, and ReZero residual
// (out = x + alpha*y).
//
// W arrives host-padded to [Npad, 296] (= the LDS row stride, so glds
// streams chunks as raw row-major copies) with Npad a multiple of 64.

#ifndef DC_SAN_MAIN
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#endif
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

using bf16 = __hip_bfloat16;

namespace {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

constexpr int BM = 128;
constexpr int K1 = 280;
constexpr int K1P = 288;
constexpr int NC = 64;
constexpr int W_STRIDE = 296;   // weight image row stride (conflict-free)
constexpr int O_STRIDE = 72;    // output chunk image stride

template <bool RELU, bool RESIDUAL>
__global__ __launch_bounds__(512, 2) void fused_linear_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w,
    const float* __restrict__ bias, const bf16* __restrict__ resid,
    bf16* __restrict__ out, int M, int N, int Npad, float alpha) {
  __shared__ __attribute__((aligned(16))) bf16 w_lds[NC * W_STRIDE];
  __shared__ __attribute__((aligned(16))) bf16 o_lds[BM * O_STRIDE];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int c = lane & 31;
  const int hi = lane >> 5;
  const int rg = wave >> 1;        // row group
  const int ch = wave & 1;         // 32-col half of the chunk
  const int m0 = blockIdx.x * BM;

  // Stage x image over w_lds+o_lds... x needs 128*296 elems = 37,888;
  // w_lds (18,944) + o_lds (9,216) = 28,160 < that, so stage x in two
  // passes: pull this wave's A-fragments directly with a strided-but-L2
  // friendly pattern instead: stage rows through w_lds half at a time.
  bf16x8 af[18];
  for (int half = 0; half < 2; ++half) {
    __syncthreads();
    // 16-B granules (37/row; W_STRIDE 296 elems = 592 B = 37 x 16 B).
    for (int idx = tid; idx < 64 * 37; idx += 512) {
      const int r = idx / 37, q4 = idx % 37;
      const int row = 64 * half + r;
      uint4 v = {};
      if (m0 + row < M && 8 * q4 + 8 <= K1) {
        v = *reinterpret_cast<const uint4*>(
            x + (size_t)(m0 + row) * K1 + 8 * q4);
      }
      *reinterpret_cast<uint4*>(&w_lds[r * W_STRIDE + 8 * q4]) = v;
    }
    __syncthreads();
    // Waves whose row group sits in this half pull their fragments.
    if ((32 * rg) / 64 == half) {
      const int r_local = 32 * rg - 64 * half + c;
#pragma unroll
      for (int s = 0; s < 18; ++s) {
        af[s] = *reinterpret_cast<const bf16x8*>(
            &w_lds[r_local * W_STRIDE + 16 * s + 8 * hi]);
      }
    }
  }
  __syncthreads();

  // Weights stream by glds (wave-uniform 1-KiB LDS chunks, per-lane
  // sources): the T14 register staging this replaces cost ~20 VGPRs and
  // held the kernel at 148 regs / 3 waves per SIMD. W arrives host-padded
  // to the LDS row stride (296), so a chunk is one raw row-major copy.
  auto issue_w = [&](int chunk) {
    const bf16* src = w + (size_t)chunk * NC * W_STRIDE;
#pragma unroll
    for (int i = 0; i < 5; ++i) {
      const int ck = wave + i * 8;
      if (ck < NC * W_STRIDE * 2 / 1024) {
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned*)(
                src + ck * 512 + lane * 8),
            (__attribute__((address_space(3))) unsigned*)(
                &w_lds[ck * 512]),
            16, 0, 0);
      }
    }
  };

  issue_w(0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  const int nchunk = Npad / NC;
  for (int chunk = 0; chunk < nchunk; ++chunk) {
    const int colt = 32 * ch;
    const int ncol = chunk * NC + colt + c;  // this lane's output col
    f32x16 acc = {};
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int s = 0; s < 18; ++s) {
      const bf16x8 bfr = *reinterpret_cast<const bf16x8*>(
          &w_lds[(colt + c) * W_STRIDE + 16 * s + 8 * hi]);
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af[s], bfr, acc,
                                                    0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    const float bv = (bias != nullptr && ncol < N) ? bias[ncol] : 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * hi + 32 * rg;
      float v = acc[r] + bv;
      if (RELU) v = v > 0.f ? v : 0.f;
      o_lds[row * O_STRIDE + colt + c] = __float2bfloat16(v);
    }
    __syncthreads();  // o_lds complete; w_lds consumed
    if (chunk + 1 < nchunk) issue_w(chunk + 1);

    // Coalesced copy-out of this 64-col chunk (+ optional residual).
    const int n0 = chunk * NC;
    for (int idx = tid; idx < BM * (NC / 8); idx += 512) {
      const int row = idx / (NC / 8), c8 = idx % (NC / 8);
      if (m0 + row >= M) continue;
      const int col = n0 + 8 * c8;
      if (col >= N) continue;
      bf16x8 v = *reinterpret_cast<const bf16x8*>(
          &o_lds[row * O_STRIDE + 8 * c8]);
      if (RESIDUAL) {
        const bf16x8 res = *reinterpret_cast<const bf16x8*>(
            resid + (size_t)(m0 + row) * K1 + col);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float rv = (float)res[j];
          const float vv = (float)v[j];
          v[j] = (__bf16)(rv + alpha * vv);
        }
      }
      if (col + 8 <= N) {
        *reinterpret_cast<bf16x8*>(
            out + (size_t)(m0 + row) * N + col) = v;
      } else {
        const unsigned short* vs =
            reinterpret_cast<const unsigned short*>(&v);
        for (int j = 0; j < 8 && col + j < N; ++j) {
          reinterpret_cast<unsigned short*>(
              out)[(size_t)(m0 + row) * N + col + j] = vs[j];
        }
      }
    }
    // Next chunk's weights must have landed (vmcnt also drains this
    // chunk's copy-out stores, which is harmless) and every wave must be
    // done with o_lds before the next epilogue rewrites it.
    if (chunk + 1 < nchunk) {
      asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
  }
}

}  // namespace

#ifndef DC_SAN_MAIN

at::Tensor fused_linear(at::Tensor x, at::Tensor w, at::Tensor bias,
                        at::Tensor resid, int64_t n_out, bool relu,
                        double alpha) {
  const bool residual = resid.defined() && resid.numel() > 0;
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16,
              "x must be bf16 on device");
  auto xc = x.contiguous();
  const int K = xc.size(-1);
  const int M = xc.numel() / K;
  TORCH_CHECK(K == K1, "fused_linear requires width-280 input");
  const int Npad = w.size(0);
  TORCH_CHECK(w.size(1) == W_STRIDE && Npad % NC == 0,
              "w must be [Npad (mult of 64), 296]");
  const int N = (int)n_out;
  TORCH_CHECK(N <= Npad, "n_out exceeds padded weight rows");
  TORCH_CHECK(!residual || N == K1, "residual requires N == 280");
  const bf16* resid_ptr = nullptr;
  at::Tensor rc;
  if (residual) {
    rc = resid.contiguous();
    TORCH_CHECK(rc.dtype() == at::kBFloat16 && rc.numel() == xc.numel(),
                "resid must match x");
    resid_ptr = reinterpret_cast<bf16*>(rc.data_ptr());
  }
  auto out = at::empty({M, N}, xc.options());
  const float* bias_ptr = nullptr;
  at::Tensor bc;
  if (bias.defined() && bias.numel() > 0) {
    bc = bias.contiguous();
    TORCH_CHECK(bc.dtype() == at::kFloat && bc.numel() >= N, "bias fp32");
    bias_ptr = bc.data_ptr<float>();
  }
  dim3 grid((M + BM - 1) / BM);
  dim3 block(512);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  auto launch = [&](auto relu_t, auto res_t) {
    hipLaunchKernelGGL(
        (fused_linear_kernel<decltype(relu_t)::value,
                             decltype(res_t)::value>),
        grid, block, 0, stream,
        reinterpret_cast<bf16*>(xc.data_ptr()),
        reinterpret_cast<bf16*>(w.data_ptr()), bias_ptr, resid_ptr,
        reinterpret_cast<bf16*>(out.data_ptr()), M, N, Npad,
        (float)alpha);
  };
  using T = std::true_type;
  using F = std::false_type;
  if (relu && residual) launch(T{}, T{});
  else if (relu) launch(T{}, F{});
  else if (residual) launch(F{}, T{});
  else launch(F{}, F{});
  return out;
}

#endif  // DC_SAN_MAIN
