// MFMA banded attention forward for gfx950 (production shape D=140, L<=104).
//
// Same math as banded_attn.hip (attention_layer.py:196-218: banded QK^T,
// fp32 softmax over |i-j| <= win, P V), restructured for the CDNA4 matrix
// cores:
//  * PERSISTENT workgroups (grid 512 = 2 blocks/CU): each block of 4
//    waves (32 query rows each) walks many (b, h) items — the one-shot
//    grid was launch/drain-bound — and prefetches the NEXT item's K/V
//    into named staging registers during the current softmax/PV;
//  * swapped QK^T — mfma(K, Q) — so each lane holds a full query row's band
//    scores and softmax is in-lane (one shfl_xor(32) pair to merge halves);
//  * P repacked to MFMA A-fragments with v_cvt_pk_bf16_f32 +
//    permlane32_swap (the cdna_hip_programming.md T12 pattern);
//  * K staged row-major in LDS (stride 152 elems) in 16-B granules, V
//    staged TRANSPOSED (Vt[dim][key], stride 168, keys shifted by win so
//    every PV fragment read is 16-byte aligned even at the negative
//    key-window of the first row block) with a key-major conflict-free
//    scatter;
//  * 38 v_mfma_f32_32x32x16_bf16 per wave per (b,h), fp32 accumulation;
//    ~79 KB LDS => 2 workgroups per CU. 626 -> 287 us at B=4096
//    (profiles/r01_perf_journal.md).
//
// Out-of-band and out-of-range keys are masked to -inf before softmax; the
// corresponding P entries are exactly 0, and the V slots they multiply are
// zero-filled, so clamped/padded reads never contribute.

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

constexpr int AM_L = 104;        // max rows (production window: 100)
constexpr int AM_D = 140;        // head dim (production: 280/2)
constexpr int K_STRIDE = 152;    // K LDS row stride (bf16), 16B-aligned rows
// Key slot shift is `win` at runtime: slot = win + key, so PV fragment
// reads land at 32w+16s+8hi (always 16-B aligned) for any win <= 12.
constexpr int V_STRIDE = 168;    // Vt row stride (key slots)
constexpr int V_ROWS = 144;      // Vt rows (dims 0..143, 140..143 zero)

__device__ __forceinline__ unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// Un-hoistable lane id (see fused_ffn_v3.hip): at 256 VGPRs the
// allocator spills lane-derived loop invariants, and each in-loop
// scratch reload's compiler vmcnt(0) drains the cross-item K/V
// prefetch — recomputing per item keeps those values loop-local.
__device__ __forceinline__ int am_lane_recompute() {
  int l;
  asm volatile(
      "v_mbcnt_lo_u32_b32 %0, -1, 0\n\t"
      "v_mbcnt_hi_u32_b32 %0, -1, %0"
      : "=v"(l));
  return l;
}

template <bool PREFETCH, bool SAVE_P = false>
__global__ __launch_bounds__(256, 2) void banded_attn_mfma_kernel(
    const bf16* __restrict__ qkv, bf16* __restrict__ out,
    int B, int L, int H, int win, float scale,
    bf16* __restrict__ p_out = nullptr,
    const uint8_t* __restrict__ drop_mask = nullptr,
    float keep_inv = 1.0f) {
  constexpr int D = AM_D;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int c = lane & 31;
  const int hi = lane >> 5;

  __shared__ __attribute__((aligned(16))) bf16 k_lds[AM_L][K_STRIDE];
  __shared__ __attribute__((aligned(16))) bf16 vt_lds[V_ROWS][V_STRIDE];

  const size_t RS = (size_t)3 * H * D;  // qkv row stride (elems)

  // One-time zero of the whole Vt image: staging below only ever writes the
  // live slots [d < D][win .. win+L), so pad rows/cols stay zero across all
  // persistent-loop items (L, win are uniform).
  for (int idx = tid; idx < V_ROWS * V_STRIDE / 8; idx += 256) {
    *(reinterpret_cast<bf16x8*>(&vt_lds[0][0]) + idx) = bf16x8{};
  }

  // K/V staging registers for the item-level prefetch pipeline: the NEXT
  // item's 16-B granules are issued during the CURRENT item's softmax/PV,
  // so their latency hides behind compute. NAMED scalars — an array here
  // would be silently promoted to LDS (AMDGPUPromoteAlloca).
  uint4 kr0, kr1, kr2, kr3, kr4, kr5, kr6, kr7;
  uint4 vr0, vr1, vr2, vr3, vr4, vr5, vr6, vr7;
  const int n_kg = L * 19;  // K granules: 19 x 16 B per row (stride 304 B)
  // V granules: 8 dims of one key each, indexed key-major so consecutive
  // lanes WRITE consecutive key slots of one Vt row (conflict-free 2-B
  // writes) — the dim-major order made lanes stride 8 Vt rows apart,
  // hitting 2 LDS banks 32-ways (PMC: conflict cycles ~= active cycles).
  // The now-uncoalesced 16-B global reads stay in L2 (104-row working
  // set) and their latency rides the cross-item prefetch.
  const int n_vg = L * 18;

  // Issue the global loads for item `it`'s K and V into the staging
  // registers. Tail granules (dims 136..139) take the 2-dword path; the
  // clamped duplicate granule on inactive threads is benign (same value
  // rewritten). Dims >= D stay zero.
  auto issue_kv = [&](int it) {
    const bf16* bs = qkv + (size_t)(it / H) * L * RS + (size_t)(it % H) * D;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int idx = min(tid + i * 256, n_kg - 1);
      const int r = idx / 19, q4 = idx % 19;
      uint4 v = {};
      const bf16* src = bs + r * RS + H * D + 8 * q4;
      if (8 * q4 + 8 <= D) {
        v = *reinterpret_cast<const uint4*>(src);
      } else if (8 * q4 < D) {
        const unsigned* p = reinterpret_cast<const unsigned*>(src);
        v.x = p[0];
        v.y = p[1];
      }
      if (i == 0) kr0 = v; else if (i == 1) kr1 = v;
      else if (i == 2) kr2 = v; else if (i == 3) kr3 = v;
      else if (i == 4) kr4 = v; else if (i == 5) kr5 = v;
      else if (i == 6) kr6 = v; else kr7 = v;
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int idx = min(tid + i * 256, n_vg - 1);
      const int r = idx % L, d0 = 8 * (idx / L);
      uint4 v = {};
      const bf16* src = bs + r * RS + 2 * H * D + d0;
      if (d0 + 8 <= D) {
        v = *reinterpret_cast<const uint4*>(src);
      } else {
        const unsigned* p = reinterpret_cast<const unsigned*>(src);
        v.x = p[0];
        v.y = p[1];
      }
      if (i == 0) vr0 = v; else if (i == 1) vr1 = v;
      else if (i == 2) vr2 = v; else if (i == 3) vr3 = v;
      else if (i == 4) vr4 = v; else if (i == 5) vr5 = v;
      else if (i == 6) vr6 = v; else vr7 = v;
    }
  };
  // Drain the staging registers into LDS. K rows are 16-B-aligned vector
  // writes; V scatters 8 ds_write_b16 down a Vt column (the tail granule's
  // zero high half rewrites pad rows 140..143 with zeros, which is their
  // required value).
  auto write_kv = [&]() {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int idx = min(tid + i * 256, n_kg - 1);
      const int r = idx / 19, q4 = idx % 19;
      *reinterpret_cast<uint4*>(&k_lds[r][8 * q4]) =
          (i == 0 ? kr0 : i == 1 ? kr1 : i == 2 ? kr2 : i == 3 ? kr3
           : i == 4 ? kr4 : i == 5 ? kr5 : i == 6 ? kr6 : kr7);
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int idx = min(tid + i * 256, n_vg - 1);
      const int r = idx % L, d0 = 8 * (idx / L);
      const uint4 v =
          (i == 0 ? vr0 : i == 1 ? vr1 : i == 2 ? vr2 : i == 3 ? vr3
           : i == 4 ? vr4 : i == 5 ? vr5 : i == 6 ? vr6 : vr7);
      const unsigned short* vs = reinterpret_cast<const unsigned short*>(&v);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        vt_lds[d0 + j][win + r] = __ushort_as_bfloat16(vs[j]);
    }
  };

  // Persistent CTA: each resident block walks many (b, h) items. The
  // one-shot version (grid = B*H tiny blocks) was launch/drain-bound: 74%
  // of wave-slots parked, ~20 us wall per ~1 us of math.
  const int BH = B * H;
  if (PREFETCH && blockIdx.x < BH) issue_kv(blockIdx.x);
  for (int item = blockIdx.x; item < BH; item += gridDim.x) {
    const int ln_ = am_lane_recompute();
    const int c = ln_ & 31;    // shadow the entry copies: loop-local,
    const int hi = ln_ >> 5;   // dead at the backedge, nothing to spill
    const bf16* base =
        qkv + (size_t)(item / H) * L * RS + (size_t)(item % H) * D;
    __syncthreads();  // prior item's LDS reads done before re-staging
    if (!PREFETCH) issue_kv(item);  // DC_ATTN_PREFETCH=0 diagnostic mode

    // ---- Issue Q B-fragment loads (this lane's query row, 9 k-steps of
    // 16) before the LDS drain so their latency hides behind it. ----
    const int l0w = 32 * wave;
    const int qrow = l0w + c;
    bf16x8 qf[9];
    {
      const bool qv = qrow < L;
      const unsigned short* qp =
          reinterpret_cast<const unsigned short*>(base + (size_t)qrow * RS);
#pragma unroll
      for (int s = 0; s < 9; ++s) {
        const int d0 = 16 * s + 8 * hi;
        bf16x8 t = {};
        if (qv) {
          if (d0 + 8 <= D) {
            t = *reinterpret_cast<const bf16x8*>(
                reinterpret_cast<const bf16*>(qp) + d0);
          } else {
            unsigned short* tw = reinterpret_cast<unsigned short*>(&t);
            for (int j = 0; j < 8; ++j) tw[j] = (d0 + j < D) ? qp[d0 + j] : 0;
          }
        }
        qf[s] = t;
      }
    }
    write_kv();
    __syncthreads();

    // ---- QK^T (swapped): S^T tiles; lane holds qrow=l0w+c, keys in regs. ----
    const int kw0 = l0w - win;  // key window start (may be negative)
    float st[32];
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      f32x16 acc = {};
      const int krow = min(max(kw0 + 32 * t + c, 0), L - 1);
#pragma unroll
      for (int s = 0; s < 9; ++s) {
        const bf16x8 a =
            *reinterpret_cast<const bf16x8*>(&k_lds[krow][16 * s + 8 * hi]);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, qf[s], acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 16; ++r) st[16 * t + r] = acc[r];
    }

    // Prefetch the next item's K/V now: the loads fly during softmax + PV
    // and are drained into LDS only after the top-of-loop barrier.
    if (PREFETCH) {
      const int next_item = item + gridDim.x;
      if (next_item < BH) issue_kv(next_item);
    }

    // ---- Masked softmax, in-lane + one half-merge. ----
    // Key-local index of st[i]: kl(i) = 32*(i/16) + (i&3) + 8*((i&15)>>2)
    // + 4*hi.
    float m = -1e30f;
#pragma unroll
    for (int i = 0; i < 32; ++i) {
      const int kl = 32 * (i >> 4) + (i & 3) + 8 * ((i & 15) >> 2) + 4 * hi;
      const int kg = kw0 + kl;
      const bool valid =
          (kl >= c) && (kl <= c + 2 * win) && (kg >= 0) && (kg < L);
      st[i] = valid ? st[i] * scale : -1e30f;
      m = fmaxf(m, st[i]);
    }
    m = fmaxf(m, __shfl_xor(m, 32, 64));
    float denom = 0.f;
#pragma unroll
    for (int i = 0; i < 32; ++i) {
      st[i] = (st[i] <= -1e29f) ? 0.f : __expf(st[i] - m);
      denom += st[i];
    }
    denom += __shfl_xor(denom, 32, 64);
    const float inv = (denom > 0.f) ? 1.f / denom : 0.f;
#pragma unroll
    for (int i = 0; i < 32; ++i) st[i] *= inv;

    if (SAVE_P) {
      // Training-forward mode: persist the band softmax (pre-dropout,
      // bf16 like the torch path's cast) as P[item][qrow][w], w = kl-c
      // in [0, 2win], then apply the caller's dropout band mask to the
      // PV input registers.
      const int W = 2 * win + 1;
      const int qr = l0w + c;
      if (qr < L) {
        bf16* prow = p_out + ((size_t)item * L + qr) * W;
        const uint8_t* mrow =
            drop_mask ? drop_mask + ((size_t)item * L + qr) * W : nullptr;
#pragma unroll
        for (int i = 0; i < 32; ++i) {
          const int kl =
              32 * (i >> 4) + (i & 3) + 8 * ((i & 15) >> 2) + 4 * hi;
          const int w = kl - c;
          if (w >= 0 && w < W) {
            const bf16 pb = __float2bfloat16(st[i]);
            prow[w] = pb;
            float pd = __bfloat162float(pb);
            if (mrow) pd *= mrow[w] ? keep_inv : 0.f;
            st[i] = pd;
          }
        }
      }
    }

    // ---- Repack P to A-fragments: 4 k-steps x 4 dwords (T12 pattern).
    // Lane pair (l, l+32) holds interleaved keys; after cvt_pk + two
    // permlane32_swaps per k-step, lane l's fragment covers its 8 contiguous
    // keys 16s+8*hi .. +7.
    bf16x8 pa[4];
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const unsigned x = cvt_pk_bf16(st[8 * s + 0], st[8 * s + 1]);
      const unsigned y = cvt_pk_bf16(st[8 * s + 2], st[8 * s + 3]);
      const unsigned x2 = cvt_pk_bf16(st[8 * s + 4], st[8 * s + 5]);
      const unsigned y2 = cvt_pk_bf16(st[8 * s + 6], st[8 * s + 7]);
      const auto rx = __builtin_amdgcn_permlane32_swap(x, x2, false, false);
      const auto ry = __builtin_amdgcn_permlane32_swap(y, y2, false, false);
      unsigned u[4] = {(unsigned)rx[0], (unsigned)ry[0], (unsigned)rx[1],
                       (unsigned)ry[1]};
      pa[s] = *reinterpret_cast<const bf16x8*>(u);
    }

    // ---- PV: out[32 qrows x 140] in 5 col tiles. ----
    // B-frag: Vt[vdim][win + kw0 + 16s + 8*hi ... +7] = Vt[vdim][32w + 16s
    // + 8*hi ...], which is >= 0 and == 0 (mod 8): reads stay 16-B aligned.
    bf16* ob =
        out + (size_t)(item / H) * L * (H * D) + (size_t)(item % H) * D;
#pragma unroll
    for (int ct = 0; ct < 5; ++ct) {
      f32x16 acc = {};
      const int vdim = min(32 * ct + c, V_ROWS - 1);
#pragma unroll
      for (int s = 0; s < 4; ++s) {
        const int kk = win + kw0 + 16 * s + 8 * hi;
        const bf16x8 bfrag =
            *reinterpret_cast<const bf16x8*>(&vt_lds[vdim][kk]);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[s], bfrag, acc,
                                                      0, 0, 0);
      }
      const int col = 32 * ct + c;
      if (col < D) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qr = l0w + (r & 3) + 8 * (r >> 2) + 4 * hi;
          if (qr < L)
            ob[(size_t)qr * H * D + col] = __float2bfloat16(acc[r]);
        }
      }
    }
  }
}

}  // namespace

#ifndef DC_SAN_MAIN

std::vector<at::Tensor> banded_attn_mfma_train_fwd(
    at::Tensor qkv, int64_t H, int64_t win, double scale,
    at::Tensor drop_mask, double p_drop) {
  TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == at::kBFloat16,
              "qkv must be bf16 on device");
  auto q = qkv.contiguous();
  const int B = q.size(0), L = q.size(1);
  const int D = q.size(2) / (3 * H);
  TORCH_CHECK(D == AM_D && L <= AM_L && L >= 32,
              "banded_attn_mfma requires D=140, 32<=L<=104");
  TORCH_CHECK(win <= 12 && win >= 1, "win must be in [1, 12]");
  const int W = 2 * (int)win + 1;
  auto out = at::empty({B, L, H * D}, q.options());
  auto p = at::zeros({B * (int)H, L, W}, q.options());
  const bool has_mask = drop_mask.defined() && drop_mask.numel() > 0;
  at::Tensor mc;
  const uint8_t* mptr = nullptr;
  if (has_mask) {
    mc = drop_mask.contiguous();
    mptr = (const uint8_t*)mc.data_ptr();
  }
  const float keep_inv =
      p_drop > 0 ? (float)(1.0 / (1.0 - p_drop)) : 1.0f;
  dim3 grid(std::min(B * (int)H, 512));
  dim3 block(256);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((banded_attn_mfma_kernel<true, true>), grid, block,
                     0, stream, reinterpret_cast<bf16*>(q.data_ptr()),
                     reinterpret_cast<bf16*>(out.data_ptr()), B, L,
                     (int)H, (int)win, (float)scale,
                     reinterpret_cast<bf16*>(p.data_ptr()), mptr,
                     keep_inv);
  return {out, p};
}

at::Tensor banded_attn_mfma(at::Tensor qkv, int64_t H, int64_t win,
                            double scale) {
  TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == at::kBFloat16,
              "qkv must be bf16 on device");
  auto q = qkv.contiguous();
  const int B = q.size(0), L = q.size(1);
  const int D = q.size(2) / (3 * H);
  TORCH_CHECK(D == AM_D && L <= AM_L && L >= 32,
              "banded_attn_mfma requires D=140, 32<=L<=104");
  TORCH_CHECK(win <= 12 && win >= 1, "win must be in [1, 12]");
  auto out = at::empty({B, L, H * D}, q.options());
  // Persistent grid: 2 blocks resident per CU (80 KB LDS each) on 256 CUs.
  dim3 grid(std::min(B * (int)H, 512));
  dim3 block(256);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  // DC_ATTN_PREFETCH=0 disables the cross-item K/V prefetch pipeline
  // (diagnostic knob for the large-batch corruption investigation).
  static const bool prefetch = [] {
    const char* e = std::getenv("DC_ATTN_PREFETCH");
    return !(e && e[0] == '0');
  }();
  if (prefetch) {
    hipLaunchKernelGGL(banded_attn_mfma_kernel<true>, grid, block, 0,
                       stream, reinterpret_cast<bf16*>(q.data_ptr()),
                       reinterpret_cast<bf16*>(out.data_ptr()),
                       B, L, (int)H, (int)win, (float)scale);
  } else {
    hipLaunchKernelGGL(banded_attn_mfma_kernel<false>, grid, block, 0,
                       stream, reinterpret_cast<bf16*>(q.data_ptr()),
                       reinterpret_cast<bf16*>(out.data_ptr()),
                       B, L, (int)H, (int)win, (float)scale);
  }
  return out;
}

#endif  // DC_SAN_MAIN
