// Fused FFN v3 for gfx950: 256-row tiles, register-resident h, glds weights.
//
// Same math as fused_ffn.hip (ffn_layer.py:69-87 + ReZero residual,
// encoder_stack.py:88-92), restructured around two measured facts:
//  * v1/v2 (BM=128) are weight-traffic bound: every block re-reads the full
//    2.3 MB of W1+W2, so doubling the row tile halves the dominant term;
//  * at 512 threads the register file (512/SIMD, 2 waves/SIMD at 254
//    VGPRs) is the binding constraint, not LDS — so h never touches LDS
//    (B1 is computed SWAPPED, mfma(W1, x), and its D result is repacked
//    in-register to B2 A-fragments with the attention kernel's T12
//    cvt_pk_bf16 + permlane32_swap pattern), and the weights stream by
//    global_load_lds (zero staging registers), double-buffered.
//
// Layout: 512 threads = 8 waves, each owning 32 of the 256 rows (no
// column split — the whole 280-wide output lives in this wave's oacc[9],
// which is what makes the register budget: 72 af + 144 oacc + ~40 working).
// Per chunk each wave runs B1 over both 32-hidden tiles of the chunk
// (2 x 18 MFMAs), repacking each tile immediately into pa[2] and running
// its two B2 k-steps (2 x 18 MFMAs) before the next tile, so only half of
// st/pa is ever live. W1 arrives in the v2 host layout [2048, 296] with b1
// folded into column 287; W2 padded [320, 2048] (rows 288+ never staged).
// LDS: 2x37,888 (W1) + 2x41,472 (W2) + 2 KiB dma-overflow scratch
// = 160,768 B of the 163,840 B/CU — one barrier per chunk.

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

constexpr int BM = 256;
constexpr int K1 = 280;
constexpr int K1P = 296;        // x-image / W1 row stride, bias col 287
constexpr int BIAS_COL = 287;
constexpr int NC = 64;
constexpr int NHID = 2048;
constexpr int NCHUNK = NHID / NC;
constexpr int W2_STRIDE = 72;   // 9 granules/row; granule 8 is a pad slot
constexpr int W2_ROWS = 288;    // staged rows (outputs 0..279 + 8 pad)
constexpr int NOUT = 280;

constexpr int W1_ELEMS = NC * K1P;          // 18,944 elems / buffer
constexpr int W2_ELEMS = W2_ROWS * W2_STRIDE;  // 20,736 elems / buffer
constexpr int W1_CHUNKS = W1_ELEMS * 2 / 1024;  // 37 KiB -> 5 issues/wave
constexpr int W2_GRAN = W2_ROWS * 9;        // 2,592 granules
constexpr int W2_CHUNKS = (W2_GRAN + 63) / 64;  // 41 -> 6 issues/wave

constexpr int OFF_W1 = 0;                       // elems; [2] buffers
constexpr int OFF_W2 = 2 * W1_ELEMS;            // 37,888
constexpr int OFF_SCRATCH = OFF_W2 + 2 * W2_ELEMS;  // 79,360
constexpr int LDS_ELEMS = OFF_SCRATCH + 1024;   // 160,768 B total

__device__ __forceinline__ void glds16(const bf16* gsrc, bf16* ldst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned*)gsrc,
      (__attribute__((address_space(3))) unsigned*)ldst, 16, 0, 0);
}

__device__ __forceinline__ unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// Lane id recomputed at the call site (volatile: un-hoistable). The
// allocator was spilling ~8 loop-invariant lane-derived LDS addresses
// at 256 VGPRs, and every in-loop scratch reload carries a compiler
// s_waitcnt vmcnt(0) that drains the in-flight weight DMA — the
// ablation probe measured the result as ZERO transfer/compute overlap
// (full == loads-only + mfma-only exactly). Two VALU per use beats a
// scratch round trip + queue drain.
__device__ __forceinline__ int lane_recompute() {
  int l;
  asm volatile(
      "v_mbcnt_lo_u32_b32 %0, -1, 0\n\t"
      "v_mbcnt_hi_u32_b32 %0, -1, %0"
      : "=v"(l));
  return l;
}

__global__ __launch_bounds__(512, 1) void fused_ffn_v3_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w1,
    const bf16* __restrict__ w2, const float* __restrict__ b2,
    bf16* __restrict__ out, int M, float alpha) {
  __shared__ __attribute__((aligned(16))) bf16 smem[LDS_ELEMS];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  // wave id lives in an SGPR (readfirstlane) — free to keep live.
  const int wave = __builtin_amdgcn_readfirstlane(tid >> 6);
  const int c = lane & 31;
  const int hi = lane >> 5;
  const int m0 = blockIdx.x * BM;

  // ---- Stage x through the W2 region (buffer 1: 20,736 >= 18,944 elems),
  // 64 rows per pass in 16-B granules; granule 35 carries the bias-column
  // constant bf16(1.0) at col 287. Waves pull their A-fragments when their
  // rows are resident. ----
  bf16x8 af[18];
  {
    bf16* ximg = &smem[OFF_W2 + W2_ELEMS];
    for (int pass = 0; pass < 4; ++pass) {
      __syncthreads();
      for (int idx = tid; idx < 64 * 37; idx += 512) {
        const int r = idx / 37, q4 = idx % 37;
        const int row = 64 * pass + r;
        uint4 v = {};
        const bool rv = (m0 + row) < M;
        if (rv && 8 * q4 + 8 <= K1) {
          v = *reinterpret_cast<const uint4*>(
              x + (size_t)(m0 + row) * K1 + 8 * q4);
        } else if (rv && 8 * q4 + 7 == BIAS_COL) {
          v.w = 0x3f800000u;  // upper half bf16(1.0) at col 287
        }
        *reinterpret_cast<uint4*>(&ximg[r * K1P + 8 * q4]) = v;
      }
      __syncthreads();
      if ((wave >> 1) == pass) {
        const int r_local = 32 * (wave & 1) + c;
#pragma unroll
        for (int s = 0; s < 18; ++s) {
          af[s] = *reinterpret_cast<const bf16x8*>(
              &ximg[r_local * K1P + 16 * s + 8 * hi]);
        }
      }
    }
  }

  // ---- glds issue helpers: wave-uniform 1-KiB LDS chunks, per-lane
  // global sources; overflow chunks land in the scratch region so every
  // wave always has 5 + 6 DMAs in flight. ----
  // Every wait below is vmcnt(0), so per-wave DMA counts need not be
  // uniform: out-of-range chunks are simply exec-masked off (a partial
  // last chunk would otherwise deposit past the buffer — glds lane
  // deposits are unconditional at dst + 16*lane).
  // Lane id re-derived per lambda and the glds LDS destination pinned
  // to an SGPR via readfirstlane — hardening against two allocator
  // hazards found in the training clone of this kernel (ffn_train.hip):
  // kernel-top lane state spilled around the chunk loop, and a spilled
  // M0 source rebuilding the DMA destination from scratch under a
  // partial exec mask (= weight deposits at arbitrary LDS offsets).
  auto issue_w1 = [&](int chunk, int buf) {
    const int ln = lane_recompute();
    const bf16* src = w1 + (size_t)chunk * W1_ELEMS;
#pragma unroll
    for (int i = 0; i < 5; ++i) {
      const int ck = wave + i * 8;
      if (ck < W1_CHUNKS) {
        const int dst_off = __builtin_amdgcn_readfirstlane(
            OFF_W1 + buf * W1_ELEMS + ck * 512);
        glds16(src + ck * 512 + ln * 8, &smem[dst_off]);
      }
    }
  };
  auto issue_w2 = [&](int chunk, int buf) {
    const int ln = lane_recompute();
#pragma unroll
    for (int i = 0; i < 6; ++i) {
      const int ck = wave + i * 8;
      const int g = ck * 64 + ln;
      if (g < W2_GRAN) {
        const int row = g / 9, sub = g % 9;
        const int k8 = sub == 8 ? 0 : sub;  // pad slot re-loads granule 0
        const int dst_off = __builtin_amdgcn_readfirstlane(
            OFF_W2 + buf * W2_ELEMS + ck * 512);
        glds16(w2 + (size_t)row * NHID + chunk * NC + 8 * k8,
               &smem[dst_off]);
      }
    }
  };

  // x staging used the W2[1] region — the first issue targeting it is
  // W2(1) below, which lands behind this barrier + the loop's waits.
  issue_w1(0, 0);
  issue_w2(0, 0);
  // lgkmcnt(0) is load-bearing: the af ds_reads issued in the staging
  // pass are lgkm-tracked, and a raw s_barrier does NOT drain counters
  // (__syncthreads would). Without it a wave can pass the barrier with
  // af reads still in flight while another wave's chunk-1 glds
  // overwrites the staging region (W2[1]) — a nondeterministic af
  // corruption measured in the dgrad instantiation (a handful of NaN
  // elements per million, schedule-dependent).
  asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  f32x16 oacc[9] = {};

  for (int chunk = 0; chunk < NCHUNK; ++chunk) {
    const int ln = lane_recompute();
    const int c = ln & 31;   // shadow the entry values: loop-local,
    const int hi = ln >> 5;  // dead at the backedge -> nothing to spill
    const int buf = chunk & 1;
    const bool more = chunk + 1 < NCHUNK;
    if (more) {
      issue_w1(chunk + 1, buf ^ 1);
      issue_w2(chunk + 1, buf ^ 1);
    }
    const bf16* w1buf = &smem[OFF_W1 + buf * W1_ELEMS];
    const bf16* w2buf = &smem[OFF_W2 + buf * W2_ELEMS];

    // ---- Per 32-hidden tile: swapped B1 (hidden in regs, m in lanes),
    // ReLU, T12 repack to pa[2], then its two B2 k-steps — only one
    // tile's st/pa live at a time. ----
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      // Single accumulator: a second k-interleaved acc (to break the
      // 18-deep dependent MFMA chain) costs 16 more live VGPRs and sent
      // the allocator from 20 to 316 spills — measured net loss.
      f32x16 acc = {};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int s = 0; s < 18; ++s) {
        const bf16x8 wfr = *reinterpret_cast<const bf16x8*>(
            &w1buf[(32 * t + c) * K1P + 16 * s + 8 * hi]);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(wfr, af[s], acc,
                                                      0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
      float st[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) st[r] = acc[r] > 0.f ? acc[r] : 0.f;
      bf16x8 pa[2];
#pragma unroll
      for (int s = 0; s < 2; ++s) {
        const unsigned x0 = cvt_pk_bf16(st[8 * s + 0], st[8 * s + 1]);
        const unsigned y0 = cvt_pk_bf16(st[8 * s + 2], st[8 * s + 3]);
        const unsigned x1 = cvt_pk_bf16(st[8 * s + 4], st[8 * s + 5]);
        const unsigned y1 = cvt_pk_bf16(st[8 * s + 6], st[8 * s + 7]);
        const auto rx = __builtin_amdgcn_permlane32_swap(x0, x1, false,
                                                         false);
        const auto ry = __builtin_amdgcn_permlane32_swap(y0, y1, false,
                                                         false);
        unsigned u[4] = {(unsigned)rx[0], (unsigned)ry[0], (unsigned)rx[1],
                         (unsigned)ry[1]};
        pa[s] = *reinterpret_cast<const bf16x8*>(u);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int s = 0; s < 2; ++s) {
#pragma unroll
        for (int ct = 0; ct < 9; ++ct) {
          const int ocol = min(32 * ct + c, NOUT - 1);
          const bf16x8 wfr = *reinterpret_cast<const bf16x8*>(
              &w2buf[ocol * W2_STRIDE + 32 * t + 16 * s + 8 * hi]);
          oacc[ct] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              pa[s], wfr, oacc[ct], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }

    // Next chunk's 11 DMA groups must have landed; raw barrier (a
    // __syncthreads would also drain nothing extra here — vmcnt(0) is
    // already required since both buffers' DMAs are the only vmem).
    if (more) {
      asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  }

  // ---- Epilogue: b2 + ReZero alpha + residual (fully unrolled). ----
  const int lne = lane_recompute();
  const int ce = lne & 31;
  const int hie = lne >> 5;
#pragma unroll
  for (int ct = 0; ct < 9; ++ct) {
    const int col = 32 * ct + ce;
    if (col >= NOUT) continue;
    const float bias = b2[col];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * hie + 32 * wave;
      if (m0 + row < M) {
        const size_t off = (size_t)(m0 + row) * K1 + col;
        const float resid = __bfloat162float(x[off]);
        out[off] = __float2bfloat16(resid + alpha * (oacc[ct][r] + bias));
      }
    }
  }
}

}  // namespace

#ifndef DC_SAN_MAIN

at::Tensor fused_ffn_v3(at::Tensor x, at::Tensor w1, at::Tensor w2,
                        at::Tensor b2, double alpha) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16,
              "x must be bf16 on device");
  auto xc = x.contiguous();
  const int K = xc.size(-1);
  const int M = xc.numel() / K;
  TORCH_CHECK(K == K1, "fused_ffn_v3 requires width 280");
  TORCH_CHECK(w1.size(0) == NHID && w1.size(1) == K1P,
              "w1 must be [2048, 296] with b1 folded into column 287");
  TORCH_CHECK(w2.size(0) == 320 && w2.size(1) == NHID,
              "w2 must be padded [320, 2048]");
  TORCH_CHECK(b2.numel() >= NOUT, "b2 must cover 280 outputs");
  auto b2c = b2.contiguous();
  TORCH_CHECK(b2c.dtype() == at::kFloat, "b2 must be fp32");
  auto out = at::empty_like(xc);
  dim3 grid((M + BM - 1) / BM);
  dim3 block(512);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(fused_ffn_v3_kernel, grid, block, 0, stream,
                     reinterpret_cast<bf16*>(xc.data_ptr()),
                     reinterpret_cast<bf16*>(w1.data_ptr()),
                     reinterpret_cast<bf16*>(w2.data_ptr()),
                     b2c.data_ptr<float>(),
                     reinterpret_cast<bf16*>(out.data_ptr()), M,
                     (float)alpha);
  return out;
}

#endif  // DC_SAN_MAIN
