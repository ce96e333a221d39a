// Training FFN pair for gfx950: fused forward and fused dgrad, both
// instances of the fused_ffn_v3 structure (256-row tiles, register-
// resident 2048-wide intermediate via swapped-B1 + T12 repack, glds
// double-buffered weight streaming).
//
// Reference semantics: ffn_layer.py:69-87 — h = relu(x@W1^T + b1);
// hd = dropout(h); y = hd@W2^T + b2 — and its autograd transpose.
//
//  * FWD  (MODE=0):  B1-role weights = W1 (+b1 folded in column 287),
//    ReLU + inverted dropout applied to the intermediate IN REGISTERS
//    (counter-based splitmix64 hash of (seed, element index) — no mask
//    tensor exists anywhere), hd persisted to global through the
//    T12-repacked pa fragments (16-B per-lane stores, sector-aligned),
//    B2-role weights = W2, epilogue adds b2. Returns y and hd.
//  * DGRAD (MODE=1): B1-role weights = W2^T, the combined relu x dropout
//    mask is RECOVERED from hd: hd > 0 iff (kept and h_pre > 0), so
//    dh_pre = dhd * (hd>0 ? 1/(1-p) : 0) — nothing extra was saved.
//    dh_pre is persisted (the hipBLASLt split-K wgrads x^T@dh_pre and
//    hd^T@dy consume it), B2-role weights = W1^T, no bias. Returns dx
//    and dh_pre.
//
// The wgrad GEMMs ([280,M]x[M,2048] with M >> N) stay on hipBLASLt —
// split-K shapes are the library's home turf; the fused win here is the
// two activation-shaped GEMM chains plus every elementwise between them
// (relu, dropout, mask-mul: 3 x 1.7 GB of round trips gone).

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
constexpr int K1P = 296;        // x-image / B1-weight row stride
constexpr int BIAS_COL = 287;
constexpr int NC = 64;
constexpr int NHID = 2048;
constexpr int NCHUNK = NHID / NC;
constexpr int W2_STRIDE = 72;
constexpr int W2_ROWS = 288;
constexpr int NOUT = 280;

constexpr int W1_ELEMS = NC * K1P;
constexpr int W2_ELEMS = W2_ROWS * W2_STRIDE;
constexpr int W1_CHUNKS = W1_ELEMS * 2 / 1024;
constexpr int W2_GRAN = W2_ROWS * 9;
constexpr int W2_CHUNKS = (W2_GRAN + 63) / 64;

constexpr int OFF_W1 = 0;
constexpr int OFF_W2 = 2 * W1_ELEMS;
constexpr int OFF_SCRATCH = OFF_W2 + 2 * W2_ELEMS;
constexpr int LDS_ELEMS = OFF_SCRATCH + 1024;

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

__device__ __forceinline__ int lane_recompute() {
  int l;
  asm volatile(
      "v_mbcnt_lo_u32_b32 %0, -1, 0\n\t"
      "v_mbcnt_hi_u32_b32 %0, -1, %0"
      : "=v"(l));
  return l;
}

// 32-bit mix (lowbias32 family) of (seed, m row, hidden idx) -> [0,1).
// Deterministic given the host-drawn seed; statistical quality is ample
// for dropout, only the forward ever evaluates it (the backward recovers
// the mask from hd>0), and staying in 32-bit keeps the chunk loop off
// the 256-VGPR cliff (a 64-bit splitmix variant spilled 266 regs).
__device__ __forceinline__ float rand01(unsigned seed, unsigned m,
                                        unsigned h) {
  unsigned z = seed + m * 0x9E3779B9u + h * 0x85EBCA6Bu;
  z ^= z >> 16;
  z *= 0x7FEB352Du;
  z ^= z >> 15;
  z *= 0x846CA68Bu;
  z ^= z >> 16;
  return (float)(z >> 8) * (1.0f / 16777216.0f);
}

// MODE 0 = forward (relu + dropout, +b2), 1 = dgrad (mask from hread),
// 2 = dgrad WITHOUT the mask (bisection probe: dh == plain dy@W2^T).
template <int MODE>
__global__ __launch_bounds__(512, 1) void ffn_train_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ w1,
    const bf16* __restrict__ w2, const float* __restrict__ b2,
    bf16* __restrict__ out, bf16* __restrict__ hsave,
    const bf16* __restrict__ hread, int M, float p, float inv_keep,
    unsigned long long seed) {
  __shared__ __attribute__((aligned(16))) bf16 smem[LDS_ELEMS];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = __builtin_amdgcn_readfirstlane(tid >> 6);
  const int c = lane & 31;
  const int hi = lane >> 5;
  const int m0 = blockIdx.x * BM;

  // Stage x (FWD) / dy (DGRAD) through the W2[1] LDS region; the bias
  // constant at col 287 is harmless for DGRAD (its B1 image keeps that
  // column zero).
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
          v.w = 0x3f800000u;
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

  // The glds LDS destination is wave-uniform and reaches the hardware
  // through M0. Force it into an SGPR with an explicit readfirstlane:
  // under register pressure (the dgrad instantiations) the allocator
  // otherwise materializes it in a VGPR, SPILLS it, and rebuilds M0 from
  // a scratch reload under a partial exec mask — when the reloading lane
  // differs from the spilling lane, M0 is garbage and the weight DMA
  // deposits at arbitrary LDS offsets (measured: nondeterministic NaN
  // corruption in MODE 1/2 while MODE 0 happened to allocate cleanly).
  // Lane id is re-derived (volatile v_mbcnt) inside each lambda and in
  // the epilogue: the kernel-top `lane`/`c`/`hi` otherwise stay live
  // across the whole chunk loop and are what the allocator spills.
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
        const int k8 = sub == 8 ? 0 : sub;
        const int dst_off = __builtin_amdgcn_readfirstlane(
            OFF_W2 + buf * W2_ELEMS + ck * 512);
        glds16(w2 + (size_t)row * NHID + chunk * NC + 8 * k8,
               &smem[dst_off]);
      }
    }
  };

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

  // Wave-uniform hd bases (m0, wave are SGPRs): the compiler keeps these
  // in scalar registers, so only a 32-bit per-lane offset is ever live
  // in the chunk loop — the 64-bit per-lane base this replaces was what
  // the allocator spilled (and in the dgrad instantiation its reloads
  // fed nondeterministic corruption; see profiles/r02_perf_journal.md).
  bf16* const hsave_w = hsave + (size_t)(m0 + 32 * wave) * NHID;
  const bf16* const hread_w =
      (MODE == 1) ? hread + (size_t)(m0 + 32 * wave) * NHID : nullptr;
  const int mrem = M - m0 - 32 * wave;  // lane c valid iff c < mrem

  f32x16 oacc[9] = {};

  for (int chunk = 0; chunk < NCHUNK; ++chunk) {
    const int ln = lane_recompute();
    const int c = ln & 31;
    const int hi = ln >> 5;
    const int buf = chunk & 1;
    const bool more = chunk + 1 < NCHUNK;
    if (more) {
      issue_w1(chunk + 1, buf ^ 1);
      issue_w2(chunk + 1, buf ^ 1);
    }
    const bf16* w1buf = &smem[OFF_W1 + buf * W1_ELEMS];
    const bf16* w2buf = &smem[OFF_W2 + buf * W2_ELEMS];

#pragma unroll
    for (int t = 0; t < 2; ++t) {
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
      __builtin_amdgcn_sched_barrier(0);  // fence: keep B1 state local
      // MFMA D-register read hazard: the cvt_pk inline asm below reads
      // acc; in MODE 0 the ~48 relu VALU ops provide the required
      // post-MFMA delay naturally, but MODE 1/2 read acc immediately
      // and measured deterministic corruption in 3 of 4 lanes (the
      // compiler's hazard recognizer does not cover asm operands
      // against the 16-pass MFMA writeback). Two s_nop 15 = 32 cycles.
      if (MODE != 0) {
        asm volatile("s_nop 15\n\ts_nop 15" ::: "memory");
      }

      // Intermediate element (m, hidden) for lane (c, hi) at index r:
      // m = m0 + (r&3) + 8*(r>>2) + 4*hi + 32*wave (the af B-operand
      // rows), hidden = 64*chunk + 32*t + c. relu/dropout applied
      // IN-PLACE on acc (a separate st[16] doubled the live range by 16
      // registers — the difference between spilling and not).
      if (MODE == 0) {
        const int ln2 = lane_recompute();
        const unsigned hidx = 64 * chunk + 32 * t + (ln2 & 31);
        const unsigned mbase = m0 + 4 * (ln2 >> 5) + 32 * wave;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float v = acc[r] > 0.f ? acc[r] : 0.f;
          if (p > 0.f) {
            const unsigned mrow = mbase + (r & 3) + 8 * (r >> 2);
            v = rand01((unsigned)seed, mrow, hidx) < p ? 0.f
                                                       : v * inv_keep;
          }
          acc[r] = v;
        }
      }
      bf16x8 pa[2];
#pragma unroll
      for (int s = 0; s < 2; ++s) {
        const unsigned x0 = cvt_pk_bf16(acc[8 * s + 0], acc[8 * s + 1]);
        const unsigned y0 = cvt_pk_bf16(acc[8 * s + 2], acc[8 * s + 3]);
        const unsigned x1 = cvt_pk_bf16(acc[8 * s + 4], acc[8 * s + 5]);
        const unsigned y1 = cvt_pk_bf16(acc[8 * s + 6], acc[8 * s + 7]);
        const auto rx = __builtin_amdgcn_permlane32_swap(x0, x1, false,
                                                         false);
        const auto ry = __builtin_amdgcn_permlane32_swap(y0, y1, false,
                                                         false);
        unsigned u[4] = {(unsigned)rx[0], (unsigned)ry[0], (unsigned)rx[1],
                         (unsigned)ry[1]};
        pa[s] = *reinterpret_cast<const bf16x8*>(u);
      }

      // In the repacked (B2 A-fragment) layout, lane (c, hi) holds row
      // (32*wave + c), k-span 64*chunk + 32*t + 16*s + 8*hi — so the
      // intermediate is persisted as one 16-B store per pa, and DGRAD's
      // mask read is the SAME two 16-B loads (not 16 scalar gathers in
      // the pre-repack layout, which cost 128 spilled VGPRs). Lane state
      // is re-derived fresh at each use site (unhoistable v_mbcnt) so no
      // address chain is live across an MFMA block — values that span
      // one get spilled, and their in-loop reloads carry vmcnt(0) waits
      // that drain the weight DMA queue (the v3 de-spill lesson).
      if (MODE == 1) {
        const int ln2 = lane_recompute();
        if ((ln2 & 31) < mrem) {
          const unsigned hoff = (unsigned)(ln2 & 31) * NHID + 64 * chunk
                                + 32 * t + 8 * (ln2 >> 5);
          // dh_pre = dhd * (hd>0 ? 1/(1-p) : 0), applied post-repack;
          // one h fragment live at a time.
          {
            const bf16x8 h0 =
                *reinterpret_cast<const bf16x8*>(hread_w + hoff);
#pragma unroll
            for (int j = 0; j < 8; ++j) {
              pa[0][j] = (float)h0[j] > 0.f
                             ? (__bf16)((float)pa[0][j] * inv_keep)
                             : (__bf16)0.f;
            }
          }
          {
            const bf16x8 h1 =
                *reinterpret_cast<const bf16x8*>(hread_w + hoff + 16);
#pragma unroll
            for (int j = 0; j < 8; ++j) {
              pa[1][j] = (float)h1[j] > 0.f
                             ? (__bf16)((float)pa[1][j] * inv_keep)
                             : (__bf16)0.f;
            }
          }
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            pa[0][j] = (__bf16)0.f;
            pa[1][j] = (__bf16)0.f;
          }
        }
      }

      __builtin_amdgcn_sched_barrier(0);  // fence: mask/repack local
      // Persist the intermediate before B2 (pa dies at its last B2
      // use instead of staying live across all 18 MFMAs for a post-B2
      // store — the difference between 0 and ~8 spilled VGPRs here).
      {
        const int ln2 = lane_recompute();
        if ((ln2 & 31) < mrem) {
          const unsigned hoff = (unsigned)(ln2 & 31) * NHID + 64 * chunk
                                + 32 * t + 8 * (ln2 >> 5);
          *reinterpret_cast<bf16x8*>(hsave_w + hoff) = pa[0];
          *reinterpret_cast<bf16x8*>(hsave_w + hoff + 16) = pa[1];
        }
      }

      __builtin_amdgcn_sched_barrier(0);  // fence: store addrs die here
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

    if (more) {
      asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  }

  // Epilogue: plain y (+b2 in FWD); the ReZero residual/post-dropout
  // stay in the wrapper (they are part of its autograd graph).
  const int lne = lane_recompute();
  const int ce = lne & 31;
  const int hie = lne >> 5;
#pragma unroll
  for (int ct = 0; ct < 9; ++ct) {
    const int col = 32 * ct + ce;
    if (col >= NOUT) continue;
    const float bias = (MODE == 0) ? b2[col] : 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * hie + 32 * wave;
      if (m0 + row < M) {
        out[(size_t)(m0 + row) * K1 + col] =
            __float2bfloat16(oacc[ct][r] + bias);
      }
    }
  }
}

}  // namespace

#ifndef DC_SAN_MAIN

std::vector<at::Tensor> ffn_train_fwd(at::Tensor x, at::Tensor w1,
                                      at::Tensor w2, at::Tensor b2,
                                      double p_drop, int64_t seed) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kBFloat16,
              "x must be bf16 on device");
  auto xc = x.contiguous();
  const int K = xc.size(-1);
  const int M = xc.numel() / K;
  TORCH_CHECK(K == K1, "ffn_train_fwd requires width 280");
  TORCH_CHECK((int64_t)M * NHID < (1ll << 31),
              "ffn_train_fwd: M too large for 32-bit hd offsets");
  TORCH_CHECK(w1.size(0) == NHID && w1.size(1) == K1P,
              "w1 must be [2048, 296] with b1 folded into column 287");
  TORCH_CHECK(w2.size(0) == 320 && w2.size(1) == NHID,
              "w2 must be padded [320, 2048]");
  auto b2c = b2.contiguous();
  TORCH_CHECK(b2c.dtype() == at::kFloat && b2c.numel() >= NOUT,
              "b2 must be fp32 [280]");
  auto y = at::empty({M, K1}, xc.options());
  auto hd = at::empty({M, NHID}, xc.options());
  const float p = (float)p_drop;
  dim3 grid((M + BM - 1) / BM);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(ffn_train_kernel<0>, grid, dim3(512), 0, stream,
                     reinterpret_cast<bf16*>(xc.data_ptr()),
                     reinterpret_cast<bf16*>(w1.data_ptr()),
                     reinterpret_cast<bf16*>(w2.data_ptr()),
                     b2c.data_ptr<float>(),
                     reinterpret_cast<bf16*>(y.data_ptr()),
                     reinterpret_cast<bf16*>(hd.data_ptr()), nullptr, M, p,
                     p < 1.f ? 1.f / (1.f - p) : 0.f,
                     (unsigned long long)seed);
  return {y, hd};
}

std::vector<at::Tensor> ffn_train_dgrad_impl(at::Tensor dy, at::Tensor hd,
                                             at::Tensor w2t, at::Tensor w1t,
                                             double p_drop, bool nomask);

std::vector<at::Tensor> ffn_train_dgrad(at::Tensor dy, at::Tensor hd,
                                        at::Tensor w2t, at::Tensor w1t,
                                        double p_drop) {
  return ffn_train_dgrad_impl(dy, hd, w2t, w1t, p_drop, false);
}

std::vector<at::Tensor> ffn_train_dgrad_nomask(at::Tensor dy, at::Tensor hd,
                                               at::Tensor w2t,
                                               at::Tensor w1t,
                                               double p_drop) {
  return ffn_train_dgrad_impl(dy, hd, w2t, w1t, p_drop, true);
}

std::vector<at::Tensor> ffn_train_dgrad_impl(at::Tensor dy, at::Tensor hd,
                                             at::Tensor w2t, at::Tensor w1t,
                                             double p_drop, bool nomask) {
  TORCH_CHECK(dy.is_cuda() && dy.dtype() == at::kBFloat16,
              "dy must be bf16 on device");
  auto dyc = dy.contiguous();
  const int K = dyc.size(-1);
  const int M = dyc.numel() / K;
  TORCH_CHECK(K == K1, "ffn_train_dgrad requires width 280");
  TORCH_CHECK((int64_t)M * NHID < (1ll << 31),
              "ffn_train_dgrad: M too large for 32-bit hd offsets");
  TORCH_CHECK(w2t.size(0) == NHID && w2t.size(1) == K1P,
              "w2t must be [2048, 296] (W2^T image, column 287 zero)");
  TORCH_CHECK(w1t.size(0) == 320 && w1t.size(1) == NHID,
              "w1t must be padded [320, 2048] (W1^T image)");
  auto hdc = hd.contiguous();
  TORCH_CHECK(hdc.dtype() == at::kBFloat16 &&
                  hdc.numel() == (int64_t)M * NHID,
              "hd must be bf16 [M, 2048]");
  auto dx = at::empty({M, K1}, dyc.options());
  auto dh = at::empty({M, NHID}, dyc.options());
  const float p = (float)p_drop;
  dim3 grid((M + BM - 1) / BM);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  auto launch = [&](auto mode_t) {
    hipLaunchKernelGGL((ffn_train_kernel<decltype(mode_t)::value>), grid,
                       dim3(512), 0, stream,
                       reinterpret_cast<bf16*>(dyc.data_ptr()),
                       reinterpret_cast<bf16*>(w2t.data_ptr()),
                       reinterpret_cast<bf16*>(w1t.data_ptr()), nullptr,
                       reinterpret_cast<bf16*>(dx.data_ptr()),
                       reinterpret_cast<bf16*>(dh.data_ptr()),
                       reinterpret_cast<bf16*>(hdc.data_ptr()), M, p,
                       p < 1.f ? 1.f / (1.f - p) : 0.f, 0ull);
  };
  if (nomask) launch(std::integral_constant<int, 2>{});
  else launch(std::integral_constant<int, 1>{});
  return {dx, dh};
}

#endif  // DC_SAN_MAIN
