// MFMA banded attention BACKWARD for gfx950 (training; D=140, L<=104).
//
// Pairs with banded_attn_mfma.hip's SAVE_P forward. All four band
// products are instances of the serving kernel's two proven MFMA
// primitives:
//   dPd = band_scores(dO, V)        (the swapped-QK^T section, q->dO,
//                                    k->V: lane holds its row's band)
//   dq  = band_apply(scale*dS, Kt)  (the PV section, P->dS, Vt->Kt)
//   dk  = band_apply(scale*dS^T, Qt)
//   dv  = band_apply(Pd^T, dOt)
// where the TRANSPOSED bands are an index remap of the dS/Pd tiles
// kept in LDS: X^T at (row=kc, key-local kl) = X[kw0+kl][c+2win-kl] —
// the same valid window as the forward, so the T12 P->A-fragment
// repack applies verbatim after an LDS gather.
//
// Softmax VJP in-lane between the two: dP = dPd*mask/(1-p);
// r = sum(dP*P) (one shfl_xor(32) half-merge like the forward's
// denominator); dS = P*(dP - r).
//
// One 256-thread block (4 waves x 32 rows) per (b,h) item on a
// persistent grid; the transpose-image region is restaged three times
// per item (Kt, Qt, dOt) with the serving kernel's key-major
// conflict-free scatter. Layouts are the PACKED serving ones
// (qkv [B,T,3HD], dout/dqkv blocks per head) so the autograd Function
// needs no transposes.

#ifndef DC_SAN_MAIN
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#endif
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace {

using bf16b = __hip_bfloat16;
typedef __bf16 bwx8 __attribute__((ext_vector_type(8)));
typedef float f32x16b __attribute__((ext_vector_type(16)));

constexpr int BW_L = 104;
constexpr int BW_D = 140;
constexpr int ROW_STRIDE = 152;   // row image stride (bf16)
constexpr int TR_STRIDE = 168;    // transpose image stride (key slots)
constexpr int TR_ROWS = 144;
constexpr int BAND_STRIDE = 28;   // dss/pds row stride (fp32 / fp32)

__device__ __forceinline__ unsigned bw_cvt_pk(float lo, float hi) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

__device__ __forceinline__ int bw_lane() {
  int l;
  asm volatile(
      "v_mbcnt_lo_u32_b32 %0, -1, 0\n\t"
      "v_mbcnt_hi_u32_b32 %0, -1, %0"
      : "=v"(l));
  return l;
}

__global__ __launch_bounds__(256) void battn_bwd_mfma_kernel(
    const bf16b* __restrict__ qkv,   // [B, L, 3*H*D]
    const bf16b* __restrict__ p_in,  // [B*H, L, W]
    const uint8_t* __restrict__ mask,  // [B*H, L, W] or null
    const bf16b* __restrict__ dout,  // [B, L, H*D]
    bf16b* __restrict__ dqkv,        // [B, L, 3*H*D]
    int B, int L, int H, int win, float scale, float keep_inv) {
  constexpr int D = BW_D;
  const int tid = threadIdx.x;
  const int wave = __builtin_amdgcn_readfirstlane(tid >> 6);

  __shared__ __attribute__((aligned(16))) bf16b row_img[BW_L][ROW_STRIDE];
  __shared__ __attribute__((aligned(16))) bf16b tr_img[TR_ROWS][TR_STRIDE];
  __shared__ float dss[BW_L][BAND_STRIDE];
  __shared__ float pds[BW_L][BAND_STRIDE];

  const size_t RS = (size_t)3 * H * D;
  const size_t OS = (size_t)H * D;
  const int W = 2 * win + 1;

  // Zero the transpose image pads once (only live slots are rewritten).
  for (int idx = tid; idx < TR_ROWS * TR_STRIDE / 8; idx += 256) {
    *(reinterpret_cast<bwx8*>(&tr_img[0][0]) + idx) = bwx8{};
  }

  // Stage `src` rows (stride srs elems) into row_img, zero-padding
  // dims >= D. 19 x 16-B granules per row.
  auto stage_rows = [&](const bf16b* src, size_t srs) {
    for (int idx = tid; idx < L * 19; idx += 256) {
      const int r = idx / 19, q4 = idx % 19;
      uint4 v = {};
      const bf16b* s = src + (size_t)r * srs + 8 * q4;
      if (8 * q4 + 8 <= D) {
        v = *reinterpret_cast<const uint4*>(s);
      } else if (8 * q4 < D) {
        const unsigned* pp = reinterpret_cast<const unsigned*>(s);
        v.x = pp[0];
        v.y = pp[1];
      }
      *reinterpret_cast<uint4*>(&row_img[r][8 * q4]) = v;
    }
  };
  // Stage `src` rows TRANSPOSED into tr_img (key-major scatter, the
  // serving kernel's conflict-free V layout: tr[d][win + key]).
  auto stage_tr = [&](const bf16b* src, size_t srs) {
    for (int idx = tid; idx < L * 18; idx += 256) {
      const int r = idx % L, d0 = 8 * (idx / L);
      uint4 v = {};
      const bf16b* s = src + (size_t)r * srs + d0;
      if (d0 + 8 <= D) {
        v = *reinterpret_cast<const uint4*>(s);
      } else {
        const unsigned* pp = reinterpret_cast<const unsigned*>(s);
        v.x = pp[0];
        v.y = pp[1];
      }
      const unsigned short* vs = reinterpret_cast<const unsigned short*>(&v);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        tr_img[d0 + j][win + r] = __ushort_as_bfloat16(vs[j]);
    }
  };

  const int BH = B * H;
  for (int item = blockIdx.x; item < BH; item += gridDim.x) {
    const int lane = bw_lane();
    const int c = lane & 31;
    const int hi = lane >> 5;
    const int h = item % H;
    const int b = item / H;
    const bf16b* qb = qkv + (size_t)b * L * RS + (size_t)h * D;
    const bf16b* kb = qb + OS;
    const bf16b* vb = qb + 2 * OS;
    const bf16b* db = dout + (size_t)b * L * OS + (size_t)h * D;
    bf16b* dqb = dqkv + (size_t)b * L * RS + (size_t)h * D;
    bf16b* dkb = dqb + OS;
    bf16b* dvb = dqb + 2 * OS;

    __syncthreads();  // prior item's reads done
    stage_rows(vb, RS);  // V rows for dPd

    // dO B-fragments (this lane's row, 9 k-steps of 16).
    const int l0w = 32 * wave;
    const int qrow = l0w + c;
    const int kw0 = l0w - win;
    bwx8 dof[9];
    {
      const bool qv = qrow < L;
      const unsigned short* qp =
          reinterpret_cast<const unsigned short*>(db + (size_t)qrow * OS);
#pragma unroll
      for (int s = 0; s < 9; ++s) {
        const int d0 = 16 * s + 8 * hi;
        bwx8 t = {};
        if (qv) {
          if (d0 + 8 <= D) {
            t = *reinterpret_cast<const bwx8*>(
                reinterpret_cast<const bf16b*>(qp) + d0);
          } else {
            unsigned short* tw = reinterpret_cast<unsigned short*>(&t);
            for (int j = 0; j < 8; ++j)
              tw[j] = (d0 + j < D) ? qp[d0 + j] : 0;
          }
        }
        dof[s] = t;
      }
    }
    __syncthreads();

    // ---- dPd = band_scores(dO, V): swapped MFMA, lane holds its
    // row's band in st[32] (the serving QK^T section). ----
    float st[32];
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      f32x16b acc = {};
      const int krow = min(max(kw0 + 32 * t + c, 0), L - 1);
#pragma unroll
      for (int s = 0; s < 9; ++s) {
        const bwx8 a = *reinterpret_cast<const bwx8*>(
            &row_img[krow][16 * s + 8 * hi]);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, dof[s], acc,
                                                      0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 16; ++r) st[16 * t + r] = acc[r];
    }

    // ---- Softmax VJP in-lane; dS*scale and Pd land in the LDS band
    // tiles for the transposed phases. ----
    {
      const bf16b* prow = p_in + ((size_t)item * L + qrow) * W;
      const uint8_t* mrow =
          mask ? mask + ((size_t)item * L + qrow) * W : nullptr;
      float pv[32], dpv[32];
      float r_part = 0.f;
#pragma unroll
      for (int i = 0; i < 32; ++i) {
        const int kl =
            32 * (i >> 4) + (i & 3) + 8 * ((i & 15) >> 2) + 4 * hi;
        const int w = kl - c;
        const int kg = kw0 + kl;
        float p_ = 0.f, dp_ = 0.f;
        if (qrow < L && w >= 0 && w < W && kg >= 0 && kg < L) {
          p_ = __bfloat162float(prow[w]);
          dp_ = st[i];
          if (mrow) dp_ *= mrow[w] ? keep_inv : 0.f;
        }
        pv[i] = p_;
        dpv[i] = dp_;
        r_part += dp_ * p_;
      }
      const float r_all = r_part + __shfl_xor(r_part, 32, 64);
#pragma unroll
      for (int i = 0; i < 32; ++i) {
        const int kl =
            32 * (i >> 4) + (i & 3) + 8 * ((i & 15) >> 2) + 4 * hi;
        const int w = kl - c;
        const float ds = pv[i] * (dpv[i] - r_all);
        st[i] = ds * scale;  // dq/dk share the scale; fold once
        if (qrow < L && w >= 0 && w < W) {
          dss[qrow][w] = st[i];
          pds[qrow][w] = pv[i] * (mrow ? (mrow[w] ? keep_inv : 0.f)
                                       : 1.f);
        }
      }
    }
    __syncthreads();  // dss/pds complete before transposed reads;
                      // row_img free for restaging

    // ---- dq = band_apply(st = scale*dS, Kt). ----
    stage_tr(kb, RS);
    __syncthreads();
    bwx8 pa[4];
    auto repack = [&](float* sv) {
#pragma unroll
      for (int s = 0; s < 4; ++s) {
        const unsigned x = bw_cvt_pk(sv[8 * s + 0], sv[8 * s + 1]);
        const unsigned y = bw_cvt_pk(sv[8 * s + 2], sv[8 * s + 3]);
        const unsigned x2 = bw_cvt_pk(sv[8 * s + 4], sv[8 * s + 5]);
        const unsigned y2 = bw_cvt_pk(sv[8 * s + 6], sv[8 * s + 7]);
        const auto rx =
            __builtin_amdgcn_permlane32_swap(x, x2, false, false);
        const auto ry =
            __builtin_amdgcn_permlane32_swap(y, y2, false, false);
        unsigned u[4] = {(unsigned)rx[0], (unsigned)ry[0],
                         (unsigned)rx[1], (unsigned)ry[1]};
        pa[s] = *reinterpret_cast<const bwx8*>(u);
      }
    };
    auto band_apply = [&](bf16b* outb, size_t ors) {
#pragma unroll
      for (int ct = 0; ct < 5; ++ct) {
        f32x16b acc = {};
        const int vdim = min(32 * ct + c, TR_ROWS - 1);
#pragma unroll
        for (int s = 0; s < 4; ++s) {
          const int kk = win + kw0 + 16 * s + 8 * hi;
          const bwx8 bfrag =
              *reinterpret_cast<const bwx8*>(&tr_img[vdim][kk]);
          acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[s], bfrag,
                                                        acc, 0, 0, 0);
        }
        const int col = 32 * ct + c;
        if (col < D) {
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int qr = l0w + (r & 3) + 8 * (r >> 2) + 4 * hi;
            if (qr < L)
              outb[(size_t)qr * ors + col] = __float2bfloat16(acc[r]);
          }
        }
      }
    };
    repack(st);
    band_apply(dqb, RS);

    // ---- dk = band_apply(scale*dS^T, Qt): transposed-band gather
    // from dss, same valid window, same repack. ----
    __syncthreads();  // tr_img reads done before restage
    stage_tr(qb, RS);
    {
      float st2[32];
#pragma unroll
      for (int i = 0; i < 32; ++i) {
        const int kl =
            32 * (i >> 4) + (i & 3) + 8 * ((i & 15) >> 2) + 4 * hi;
        const int lrow = kw0 + kl;      // source row l
        const int w = c + 2 * win - kl;  // dS[l][w] == dS^T[kc][kl]
        float v = 0.f;
        if (qrow < L && lrow >= 0 && lrow < L && w >= 0 && w < W) {
          v = dss[lrow][w];
        }
        st2[i] = v;
      }
      __syncthreads();
      repack(st2);
    }
    band_apply(dkb, RS);

    // ---- dv = band_apply(Pd^T, dOt). ----
    __syncthreads();
    stage_tr(db, OS);
    {
      float st3[32];
#pragma unroll
      for (int i = 0; i < 32; ++i) {
        const int kl =
            32 * (i >> 4) + (i & 3) + 8 * ((i & 15) >> 2) + 4 * hi;
        const int lrow = kw0 + kl;
        const int w = c + 2 * win - kl;
        float v = 0.f;
        if (qrow < L && lrow >= 0 && lrow < L && w >= 0 && w < W) {
          v = pds[lrow][w];
        }
        st3[i] = v;
      }
      __syncthreads();
      repack(st3);
    }
    band_apply(dvb, RS);
  }
}

}  // namespace

#ifndef DC_SAN_MAIN

at::Tensor banded_attn_bwd_mfma(
    at::Tensor qkv, at::Tensor p, at::Tensor mask, at::Tensor dout,
    int64_t H, int64_t win, double p_drop) {
  TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == at::kBFloat16,
              "qkv bf16 cuda");
  auto qc = qkv.contiguous();
  auto pc = p.contiguous();
  auto dc = dout.contiguous();
  const int B = qc.size(0), L = qc.size(1);
  const int D = qc.size(2) / (3 * (int)H);
  TORCH_CHECK(D == BW_D && L <= BW_L && L >= 32,
              "banded_attn_bwd_mfma requires D=140, 32<=L<=104");
  TORCH_CHECK(win <= 12 && win >= 1, "win in [1,12]");
  const bool has_mask = mask.defined() && mask.numel() > 0;
  at::Tensor mc;
  const uint8_t* mptr = nullptr;
  if (has_mask) {
    mc = mask.contiguous();
    mptr = (const uint8_t*)mc.data_ptr();
  }
  auto dqkv = at::empty_like(qc);
  const float keep_inv =
      p_drop > 0 ? (float)(1.0 / (1.0 - p_drop)) : 1.0f;
  dim3 grid(std::min(B * (int)H, 1024));
  dim3 block(256);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(battn_bwd_mfma_kernel, grid, block, 0, stream,
                     reinterpret_cast<bf16b*>(qc.data_ptr()),
                     reinterpret_cast<bf16b*>(pc.data_ptr()), mptr,
                     reinterpret_cast<bf16b*>(dc.data_ptr()),
                     reinterpret_cast<bf16b*>(dqkv.data_ptr()), B, L,
                     (int)H, (int)win,
                     (float)(1.0 / std::sqrt((double)D)), keep_inv);
  return dqkv;
}

#endif  // DC_SAN_MAIN
