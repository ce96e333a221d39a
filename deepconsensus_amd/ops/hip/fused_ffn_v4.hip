// Fused FFN v4 for gfx950: v3 with the B1 GEMM on 16x16x32 MFMAs.
//
// v3's B1 is one 18-deep DEPENDENT mfma_32x32x16 chain per 32-hidden
// tile (single f32x16 accumulator; a second one spills at 256 VGPRs).
// v4 computes the same tile as a 2x2 grid of 16x16x32 MFMAs — four
// INDEPENDENT f32x4 accumulator chains (9 deep) at the SAME register
// cost (4x4 = 16 regs), so the MFMA pipe sees ILP 4 instead of 1.
// The T12 repack to B2's 32x32 A-fragments becomes a per-s
// v_permlane16_swap pair: one swap delivers both target dwords
// (own-tile dword to the partner 16-group, partner-tile dword back).
// Everything else (glds double buffer, x staging, epilogue, B2)
// matches fused_ffn_v3.hip.
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

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

using bf16 = __hip_bfloat16;

namespace {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

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

__global__ __launch_bounds__(512, 1) void fused_ffn_v4_kernel(
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
        // B-fragments for mfma_f32_16x16x32_bf16: af[ks*2 + j16] holds
        // B[k = ks*32 + (lane>>4)*8 + e][m = j16*16 + (lane&15)] — the
        // x row is m, the 16-B column run starts at ks*32 + (lane>>4)*8.
        const int q = lane & 15;
        const int kg = lane >> 4;  // 0..3: k-subgroup
#pragma unroll
        for (int ks = 0; ks < 9; ++ks) {
#pragma unroll
          for (int j16 = 0; j16 < 2; ++j16) {
            const int r_local = 32 * (wave & 1) + 16 * j16 + q;
            af[ks * 2 + j16] = *reinterpret_cast<const bf16x8*>(
                &ximg[r_local * K1P + 32 * ks + 8 * kg]);
          }
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
  auto issue_w1 = [&](int chunk, int buf) {
    const bf16* src = w1 + (size_t)chunk * W1_ELEMS;
    bf16* dst = &smem[OFF_W1 + buf * W1_ELEMS];
#pragma unroll
    for (int i = 0; i < 5; ++i) {
      const int ck = wave + i * 8;
      if (ck < W1_CHUNKS) glds16(src + ck * 512 + lane * 8, dst + ck * 512);
    }
  };
  auto issue_w2 = [&](int chunk, int buf) {
    bf16* dst0 = &smem[OFF_W2 + buf * W2_ELEMS];
#pragma unroll
    for (int i = 0; i < 6; ++i) {
      const int ck = wave + i * 8;
      const int g = ck * 64 + lane;
      if (g < W2_GRAN) {
        const int row = g / 9, sub = g % 9;
        const int k8 = sub == 8 ? 0 : sub;  // pad slot re-loads granule 0
        glds16(w2 + (size_t)row * NHID + chunk * NC + 8 * k8,
               dst0 + ck * 512);
      }
    }
  };

  // x staging used the W2[1] region — the first issue targeting it is
  // W2(1) below, which lands behind this barrier + the loop's waits.
  issue_w1(0, 0);
  issue_w2(0, 0);
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
      // 2x2 grid of 16x16x32 MFMAs: A = W1 rows (16 hidden x 32 k per
      // fragment, lane: i = lane&15, k = (lane>>4)*8 + e), B = af.
      // Four independent 9-deep accumulator chains (ILP 4).
      const int q = ln & 15;
      const int kg = ln >> 4;
      f32x4 a00 = {}, a01 = {}, a10 = {}, a11 = {};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < 9; ++ks) {
        const bf16x8 w0 = *reinterpret_cast<const bf16x8*>(
            &w1buf[(32 * t + q) * K1P + 32 * ks + 8 * kg]);
        const bf16x8 w1f = *reinterpret_cast<const bf16x8*>(
            &w1buf[(32 * t + 16 + q) * K1P + 32 * ks + 8 * kg]);
        a00 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            w0, af[ks * 2 + 0], a00, 0, 0, 0);
        a01 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            w0, af[ks * 2 + 1], a01, 0, 0, 0);
        a10 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            w1f, af[ks * 2 + 0], a10, 0, 0, 0);
        a11 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            w1f, af[ks * 2 + 1], a11, 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
      // ReLU + repack to B2's 32x32 A-fragments: per s (= i16), pack
      // the two j16 tiles' reg pairs and swap 16-groups once — the
      // swap's two results are exactly target dwords (d0,d2)/(d1,d3).
      bf16x8 pa[2];
#pragma unroll
      for (int s = 0; s < 2; ++s) {
        const f32x4 t0 = s ? a10 : a00;  // tile (i16=s, j16=0)
        const f32x4 t1 = s ? a11 : a01;  // tile (i16=s, j16=1)
        const unsigned c00 = cvt_pk_bf16(fmaxf(t0[0], 0.f),
                                         fmaxf(t0[1], 0.f));
        const unsigned c01 = cvt_pk_bf16(fmaxf(t0[2], 0.f),
                                         fmaxf(t0[3], 0.f));
        const unsigned c10 = cvt_pk_bf16(fmaxf(t1[0], 0.f),
                                         fmaxf(t1[1], 0.f));
        const unsigned c11 = cvt_pk_bf16(fmaxf(t1[2], 0.f),
                                         fmaxf(t1[3], 0.f));
        const auto r0 = __builtin_amdgcn_permlane16_swap(c00, c10,
                                                         false, false);
        const auto r1 = __builtin_amdgcn_permlane16_swap(c01, c11,
                                                         false, false);
        unsigned u[4] = {(unsigned)r0[0], (unsigned)r1[0],
                         (unsigned)r0[1], (unsigned)r1[1]};
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
#pragma unroll
  for (int ct = 0; ct < 9; ++ct) {
    const int col = 32 * ct + c;
    if (col >= NOUT) continue;
    const float bias = b2[col];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * hi + 32 * wave;
      if (m0 + row < M) {
        const size_t off = (size_t)(m0 + row) * K1 + col;
        const float resid = __bfloat162float(x[off]);
        out[off] = __float2bfloat16(resid + alpha * (oacc[ct][r] + bias));
      }
    }
  }
}

}  // namespace

at::Tensor fused_ffn_v4(at::Tensor x, at::Tensor w1, at::Tensor w2,
                        at::Tensor b2, double alpha) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16,
              "x must be bf16 on device");
  auto xc = x.contiguous();
  const int K = xc.size(-1);
  const int M = xc.numel() / K;
  TORCH_CHECK(K == K1, "fused_ffn_v4 requires width 280");
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
  hipLaunchKernelGGL(fused_ffn_v4_kernel, grid, block, 0, stream,
                     reinterpret_cast<bf16*>(xc.data_ptr()),
                     reinterpret_cast<bf16*>(w1.data_ptr()),
                     reinterpret_cast<bf16*>(w2.data_ptr()),
                     b2c.data_ptr<float>(),
                     reinterpret_cast<bf16*>(out.data_ptr()), M,
                     (float)alpha);
  return out;
}
