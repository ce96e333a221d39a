// Training banded attention (fwd + bwd) for gfx950.
//
// The training path previously ran the torch chain (full [T,T] matmuls
// + mask + softmax + dropout + PV, and autograd's mirror of all of it
// backward) — 4x the band's FLOPs plus a long elementwise tail
// (profiles/r01_train_top_kernels.txt). These kernels compute the
// banded form directly, matching models/model.py SelfAttention
// semantics exactly: S = (q * d^-0.5) @ k^T band-masked (-1e9 ==
// exact-zero weights in fp32 softmax), P = softmax_fp32(S) cast bf16,
// Pd = dropout(P) via a caller-supplied band mask, ctx = Pd @ v.
//
// Backward (band-local, fp32 accumulation):
//   dPd = dO @ v^T;  dP = dPd * mask / (1-p)
//   dS  = P * (dP - rowsum(dP * P))          [softmax VJP]
//   dq  = scale * dS @ k;  dk = scale * dS^T @ q;  dv = Pd^T @ dO
//
// v1 is VALU-based (the band GEMMs are [100 x 25 x 140] — tiny per
// item); one 256-thread block per (b, h), q/k/v(/dO) staged in LDS.
// Layouts: q,k,v,ctx,dO [B,H,T,D] bf16 contiguous; P [B,H,T,W] bf16;
// mask [B,H,T,W] uint8 (1 = keep).

#ifndef DC_SAN_MAIN
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#endif
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace {

using bf16_t = __hip_bfloat16;
typedef __bf16 btx8 __attribute__((ext_vector_type(8)));

// 8-wide bf16 dot-product step (G13: never scalar bf16 LDS reads).
__device__ __forceinline__ float dot8(const bf16_t* a, const bf16_t* b) {
  const btx8 av = *reinterpret_cast<const btx8*>(a);
  const btx8 bv = *reinterpret_cast<const btx8*>(b);
  float acc = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    acc += (float)av[j] * (float)bv[j];
  }
  return acc;
}

constexpr int BT_MAXT = 104;   // max window length
constexpr int BT_MAXW = 25;    // max band width (2*12+1)
constexpr int BT_MAXD = 144;   // padded head dim (D = 140)

__global__ __launch_bounds__(256) void battn_train_fwd_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, const uint8_t* __restrict__ mask,
    bf16_t* __restrict__ ctx, bf16_t* __restrict__ p_out,
    int BH, int T, int D, int win, float scale, float keep_inv) {
  __shared__ bf16_t qs[BT_MAXT][BT_MAXD];
  __shared__ bf16_t ks[BT_MAXT][BT_MAXD];
  __shared__ bf16_t vs[BT_MAXT][BT_MAXD];
  __shared__ float ps[BT_MAXT][BT_MAXW];

  const int item = blockIdx.x;
  const int tid = threadIdx.x;
  const int W = 2 * win + 1;
  const size_t base = (size_t)item * T * D;

  for (int idx = tid; idx < T * BT_MAXD; idx += 256) {
    const int t = idx / BT_MAXD, d = idx % BT_MAXD;
    const bf16_t z = __float2bfloat16(0.f);
    const bool in = d < D;
    qs[t][d] = in ? q[base + (size_t)t * D + d] : z;
    ks[t][d] = in ? k[base + (size_t)t * D + d] : z;
    vs[t][d] = in ? v[base + (size_t)t * D + d] : z;
  }
  __syncthreads();

  // S (scaled, fp32) into ps.
  for (int cell = tid; cell < T * W; cell += 256) {
    const int l = cell / W, w = cell % W;
    const int kc = l - win + w;
    float s = -1e30f;
    if (kc >= 0 && kc < T) {
      float acc = 0.f;
      for (int d = 0; d < BT_MAXD; d += 8) {
        acc += dot8(&qs[l][d], &ks[kc][d]);
      }
      s = acc * scale;
    }
    ps[l][w] = s;
  }
  __syncthreads();

  // Row softmax (fp32) -> bf16 P; one thread per row.
  for (int l = tid; l < T; l += 256) {
    float mx = -1e30f;
    for (int w = 0; w < W; ++w) mx = fmaxf(mx, ps[l][w]);
    float denom = 0.f;
    for (int w = 0; w < W; ++w) {
      const float e = (ps[l][w] <= -1e29f) ? 0.f : __expf(ps[l][w] - mx);
      ps[l][w] = e;
      denom += e;
    }
    const float inv = 1.f / denom;
    for (int w = 0; w < W; ++w) {
      // Match torch: softmax fp32 -> cast bf16 (P saved pre-dropout).
      const bf16_t pb = __float2bfloat16(ps[l][w] * inv);
      p_out[((size_t)item * T + l) * W + w] = pb;
      // Pd in fp32 from the bf16 value (torch drops out the bf16 cast).
      float pd = __bfloat162float(pb);
      if (mask != nullptr) {
        pd *= mask[((size_t)item * T + l) * W + w] ? keep_inv : 0.f;
      }
      ps[l][w] = pd;
    }
  }
  __syncthreads();

  // ctx = Pd @ v: each thread an 8-wide d-granule of one row.
  const int ng = T * (BT_MAXD / 8);
  for (int g = tid; g < ng; g += 256) {
    const int l = g / (BT_MAXD / 8), d0 = (g % (BT_MAXD / 8)) * 8;
    if (d0 >= D) continue;
    float acc[8] = {};
    const int w0 = max(0, win - l);
    const int w1 = min(W, T + win - l);
    for (int w = w0; w < w1; ++w) {
      const float pw = ps[l][w];
      const btx8 vv = *reinterpret_cast<const btx8*>(
          &vs[l - win + w][d0]);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += pw * (float)vv[j];
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (d0 + j < D) {
        ctx[base + (size_t)l * D + d0 + j] = __float2bfloat16(acc[j]);
      }
    }
  }
}

__global__ __launch_bounds__(256) void battn_train_bwd_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
    const bf16_t* __restrict__ v, const bf16_t* __restrict__ p_in,
    const uint8_t* __restrict__ mask, const bf16_t* __restrict__ dout,
    bf16_t* __restrict__ dq, bf16_t* __restrict__ dk,
    bf16_t* __restrict__ dv,
    int BH, int T, int D, int win, float scale, float keep_inv) {
  __shared__ bf16_t qs[BT_MAXT][BT_MAXD];
  __shared__ bf16_t ks[BT_MAXT][BT_MAXD];
  __shared__ bf16_t vs[BT_MAXT][BT_MAXD];
  __shared__ bf16_t dos[BT_MAXT][BT_MAXD];
  __shared__ float dss[BT_MAXT][BT_MAXW];   // dS
  __shared__ float pds[BT_MAXT][BT_MAXW];   // Pd

  const int item = blockIdx.x;
  const int tid = threadIdx.x;
  const int W = 2 * win + 1;
  const size_t base = (size_t)item * T * D;

  for (int idx = tid; idx < T * BT_MAXD; idx += 256) {
    const int t = idx / BT_MAXD, d = idx % BT_MAXD;
    const bf16_t z = __float2bfloat16(0.f);
    const bool in = d < D;
    qs[t][d] = in ? q[base + (size_t)t * D + d] : z;
    ks[t][d] = in ? k[base + (size_t)t * D + d] : z;
    vs[t][d] = in ? v[base + (size_t)t * D + d] : z;
    dos[t][d] = in ? dout[base + (size_t)t * D + d] : z;
  }
  __syncthreads();

  // dPd = dO @ v^T (band) and Pd staged.
  for (int cell = tid; cell < T * W; cell += 256) {
    const int l = cell / W, w = cell % W;
    const int kc = l - win + w;
    float dpd = 0.f, pd = 0.f;
    if (kc >= 0 && kc < T) {
      float acc = 0.f;
      for (int d = 0; d < BT_MAXD; d += 8) {
        acc += dot8(&dos[l][d], &vs[kc][d]);
      }
      dpd = acc;
      pd = __bfloat162float(p_in[((size_t)item * T + l) * W + w]);
      if (mask != nullptr) {
        pd *= mask[((size_t)item * T + l) * W + w] ? keep_inv : 0.f;
      }
    }
    pds[l][w] = pd;
    dss[l][w] = dpd;  // holds dPd for now
  }
  __syncthreads();

  // dS = P * (dP - rowsum(dP * P)); one thread per row.
  for (int l = tid; l < T; l += 256) {
    float dp_row[BT_MAXW];
    float r = 0.f;
    for (int w = 0; w < W; ++w) {
      const int kc = l - win + w;
      float p_ = 0.f, dp_ = 0.f;
      if (kc >= 0 && kc < T) {
        p_ = __bfloat162float(p_in[((size_t)item * T + l) * W + w]);
        dp_ = dss[l][w];
        if (mask != nullptr) {
          dp_ *= mask[((size_t)item * T + l) * W + w] ? keep_inv : 0.f;
        }
      }
      dp_row[w] = dp_;
      r += dp_ * p_;
    }
    for (int w = 0; w < W; ++w) {
      const int kc = l - win + w;
      float p_ = 0.f;
      if (kc >= 0 && kc < T) {
        p_ = __bfloat162float(p_in[((size_t)item * T + l) * W + w]);
      }
      dss[l][w] = p_ * (dp_row[w] - r);
    }
  }
  __syncthreads();

  // dq[l,d] = scale * sum_w dS[l,w] * k[kc,d]
  // dk[kc,d] = scale * sum_l dS[l,w(l,kc)] * q[l,d]   (w = kc - l + win)
  // dv[kc,d] = sum_l Pd[l,w] * dO[l,d]
  const int ng = T * (BT_MAXD / 8);
  for (int g = tid; g < ng; g += 256) {
    const int l = g / (BT_MAXD / 8), d0 = (g % (BT_MAXD / 8)) * 8;
    if (d0 >= D) continue;
    float aq[8] = {}, ak[8] = {}, av[8] = {};
    const int w0 = max(0, win - l);
    const int w1 = min(W, T + win - l);
    for (int w = w0; w < w1; ++w) {
      const float dsw = dss[l][w];
      const btx8 kv = *reinterpret_cast<const btx8*>(
          &ks[l - win + w][d0]);
#pragma unroll
      for (int j = 0; j < 8; ++j) aq[j] += dsw * (float)kv[j];
    }
    // Column kc = l: rows l2 within the band of kc.
    const int kc = l;
    const int l2_0 = max(0, kc - win);
    const int l2_1 = min(T, kc + win + 1);
    for (int l2 = l2_0; l2 < l2_1; ++l2) {
      const int w = kc - l2 + win;
      const float dsw = dss[l2][w];
      const float pdw = pds[l2][w];
      const btx8 qv = *reinterpret_cast<const btx8*>(&qs[l2][d0]);
      const btx8 dov = *reinterpret_cast<const btx8*>(&dos[l2][d0]);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        ak[j] += dsw * (float)qv[j];
        av[j] += pdw * (float)dov[j];
      }
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (d0 + j < D) {
        const size_t off = base + (size_t)l * D + d0 + j;
        dq[off] = __float2bfloat16(aq[j] * scale);
        dk[off] = __float2bfloat16(ak[j] * scale);
        dv[off] = __float2bfloat16(av[j]);
      }
    }
  }
}

}  // namespace

#ifndef DC_SAN_MAIN

std::vector<at::Tensor> banded_attn_train_fwd(
    at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor mask,
    int64_t win, double p_drop) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == at::kBFloat16, "q bf16 cuda");
  auto qc = q.contiguous();
  auto kc_ = k.contiguous();
  auto vc = v.contiguous();
  const int B = qc.size(0), H = qc.size(1), T = qc.size(2),
            D = qc.size(3);
  const int W = 2 * (int)win + 1;
  TORCH_CHECK(T <= BT_MAXT && D <= BT_MAXD && W <= BT_MAXW,
              "shape exceeds kernel limits");
  const bool has_mask = mask.defined() && mask.numel() > 0;
  at::Tensor mc;
  const uint8_t* mptr = nullptr;
  if (has_mask) {
    mc = mask.contiguous();
    TORCH_CHECK(mc.dtype() == at::kBool || mc.dtype() == at::kByte,
                "mask bool/byte");
    mptr = (const uint8_t*)mc.data_ptr();
  }
  auto ctx = at::empty_like(qc);
  auto p = at::empty({B, H, T, W}, qc.options());
  const float keep_inv =
      p_drop > 0 ? (float)(1.0 / (1.0 - p_drop)) : 1.0f;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(battn_train_fwd_kernel, dim3(B * H), dim3(256), 0,
                     stream, (const bf16_t*)qc.data_ptr(),
                     (const bf16_t*)kc_.data_ptr(),
                     (const bf16_t*)vc.data_ptr(), mptr,
                     (bf16_t*)ctx.data_ptr(), (bf16_t*)p.data_ptr(),
                     B * H, T, D, (int)win,
                     (float)(1.0 / std::sqrt((double)D)), keep_inv);
  return {ctx, p};
}

std::vector<at::Tensor> banded_attn_train_bwd(
    at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor p,
    at::Tensor mask, at::Tensor dout, int64_t win, double p_drop) {
  auto qc = q.contiguous();
  auto kc_ = k.contiguous();
  auto vc = v.contiguous();
  auto pc = p.contiguous();
  auto dc = dout.contiguous();
  const int B = qc.size(0), H = qc.size(1), T = qc.size(2),
            D = qc.size(3);
  const bool has_mask = mask.defined() && mask.numel() > 0;
  at::Tensor mc;
  const uint8_t* mptr = nullptr;
  if (has_mask) {
    mc = mask.contiguous();
    mptr = (const uint8_t*)mc.data_ptr();
  }
  auto dq = at::empty_like(qc);
  auto dk = at::empty_like(qc);
  auto dv = at::empty_like(qc);
  const float keep_inv =
      p_drop > 0 ? (float)(1.0 / (1.0 - p_drop)) : 1.0f;
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(battn_train_bwd_kernel, dim3(B * H), dim3(256), 0,
                     stream, (const bf16_t*)qc.data_ptr(),
                     (const bf16_t*)kc_.data_ptr(),
                     (const bf16_t*)vc.data_ptr(),
                     (const bf16_t*)pc.data_ptr(), mptr,
                     (const bf16_t*)dc.data_ptr(),
                     (bf16_t*)dq.data_ptr(), (bf16_t*)dk.data_ptr(),
                     (bf16_t*)dv.data_ptr(), B * H, T, D, (int)win,
                     (float)(1.0 / std::sqrt((double)D)), keep_inv);
  return {dq, dk, dv};
}

#endif  // DC_SAN_MAIN

// ---------------------------------------------------------------------------
// bwd v2: packed layouts + occupancy diet. Pairs with the MFMA serving
// forward's SAVE_P variant (banded_attn_mfma.hip): qkv/dout arrive in
// the serving layout ([B,L,3HD] / [B,L,HD]) so no transposes or
// contiguous() copies sit on the train path, and dq/dk/dv write one
// packed dqkv whose grad flows through a single fused QKV Linear.
// LDS holds only V + dO tiles (+ band dS fp32 / Pd bf16): K and Q rows
// stream from L2 in the 8-wide inner loops, cutting LDS from 140 KB to
// ~75 KB = 2 blocks/CU (the v1 kernel's 1-block occupancy was the
// measured 0.62x loss vs the torch chain).
// ---------------------------------------------------------------------------

namespace {

__global__ __launch_bounds__(256, 2) void battn_bwd2_kernel(
    const bf16_t* __restrict__ qkv,   // [B, T, 3*H*D]
    const bf16_t* __restrict__ p_in,  // [B*H, T, W]
    const uint8_t* __restrict__ mask, // [B*H, T, W] or null
    const bf16_t* __restrict__ dout,  // [B, T, H*D]
    bf16_t* __restrict__ dqkv,        // [B, T, 3*H*D]
    int BH, int H, int T, int D, int win, float scale, float keep_inv) {
  __shared__ bf16_t vs[BT_MAXT][BT_MAXD];
  __shared__ bf16_t dos[BT_MAXT][BT_MAXD];
  __shared__ float dss[BT_MAXT][BT_MAXW];
  __shared__ bf16_t pds[BT_MAXT][BT_MAXW];

  const int item = blockIdx.x;
  const int tid = threadIdx.x;
  const int W = 2 * win + 1;
  const int h = item % H;
  const int b = item / H;
  const size_t RS = (size_t)3 * H * D;   // qkv row stride
  const size_t OS = (size_t)H * D;       // dout/dqkv-head row stride
  const bf16_t* qb = qkv + (size_t)b * T * RS + (size_t)h * D;
  const bf16_t* kb = qb + OS;
  const bf16_t* vb = qb + 2 * OS;
  const bf16_t* db = dout + (size_t)b * T * OS + (size_t)h * D;
  bf16_t* dqb = dqkv + (size_t)b * T * RS + (size_t)h * D;
  bf16_t* dkb = dqb + OS;
  bf16_t* dvb = dqb + 2 * OS;

  for (int idx = tid; idx < T * BT_MAXD; idx += 256) {
    const int t = idx / BT_MAXD, d = idx % BT_MAXD;
    const bf16_t z = __float2bfloat16(0.f);
    const bool in = d < D;
    vs[t][d] = in ? vb[(size_t)t * RS + d] : z;
    dos[t][d] = in ? db[(size_t)t * OS + d] : z;
  }
  __syncthreads();

  // dPd = dO @ v^T (band); Pd staged bf16.
  for (int cell = tid; cell < T * W; cell += 256) {
    const int l = cell / W, w = cell % W;
    const int kc = l - win + w;
    float dpd = 0.f, pd = 0.f;
    if (kc >= 0 && kc < T) {
      float acc = 0.f;
      for (int d = 0; d < BT_MAXD; d += 8) {
        acc += dot8(&dos[l][d], &vs[kc][d]);
      }
      dpd = acc;
      pd = __bfloat162float(p_in[((size_t)item * T + l) * W + w]);
      if (mask != nullptr) {
        pd *= mask[((size_t)item * T + l) * W + w] ? keep_inv : 0.f;
      }
    }
    pds[l][w] = __float2bfloat16(pd);
    dss[l][w] = dpd;
  }
  __syncthreads();

  // dS = P * (dP - rowsum(dP * P)); one thread per row.
  for (int l = tid; l < T; l += 256) {
    float dp_row[BT_MAXW];
    float r = 0.f;
    for (int w = 0; w < W; ++w) {
      const int kc = l - win + w;
      float p_ = 0.f, dp_ = 0.f;
      if (kc >= 0 && kc < T) {
        p_ = __bfloat162float(p_in[((size_t)item * T + l) * W + w]);
        dp_ = dss[l][w];
        if (mask != nullptr) {
          dp_ *= mask[((size_t)item * T + l) * W + w] ? keep_inv : 0.f;
        }
      }
      dp_row[w] = dp_;
      r += dp_ * p_;
    }
    for (int w = 0; w < W; ++w) {
      const int kc = l - win + w;
      float p_ = 0.f;
      if (kc >= 0 && kc < T) {
        p_ = __bfloat162float(p_in[((size_t)item * T + l) * W + w]);
      }
      dss[l][w] = p_ * (dp_row[w] - r);
    }
  }
  __syncthreads();

  // Outputs: 8-wide d-granules; K and Q rows stream from L2.
  const int ng = T * (BT_MAXD / 8);
  for (int g = tid; g < ng; g += 256) {
    const int l = g / (BT_MAXD / 8), d0 = (g % (BT_MAXD / 8)) * 8;
    if (d0 >= D) continue;
    const bool full8 = d0 + 8 <= D;
    float aq[8] = {}, ak[8] = {}, av[8] = {};
    const int w0 = max(0, win - l);
    const int w1 = min(W, T + win - l);
    for (int w = w0; w < w1; ++w) {
      const float dsw = dss[l][w];
      const bf16_t* kr = kb + (size_t)(l - win + w) * RS + d0;
      if (full8) {
        const btx8 kv = *reinterpret_cast<const btx8*>(kr);
#pragma unroll
        for (int j = 0; j < 8; ++j) aq[j] += dsw * (float)kv[j];
      } else {
        for (int j = 0; j + d0 < D; ++j) {
          aq[j] += dsw * __bfloat162float(kr[j]);
        }
      }
    }
    const int kc = l;
    const int l2_0 = max(0, kc - win);
    const int l2_1 = min(T, kc + win + 1);
    for (int l2 = l2_0; l2 < l2_1; ++l2) {
      const int w = kc - l2 + win;
      const float dsw = dss[l2][w];
      const float pdw = __bfloat162float(pds[l2][w]);
      const bf16_t* qr = qb + (size_t)l2 * RS + d0;
      const btx8 dov = *reinterpret_cast<const btx8*>(&dos[l2][d0]);
      if (full8) {
        const btx8 qv = *reinterpret_cast<const btx8*>(qr);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          ak[j] += dsw * (float)qv[j];
          av[j] += pdw * (float)dov[j];
        }
      } else {
        for (int j = 0; j + d0 < D; ++j) {
          ak[j] += dsw * __bfloat162float(qr[j]);
          av[j] += pdw * (float)dov[j];
        }
      }
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (d0 + j < D) {
        const size_t off = (size_t)l * RS + d0 + j;
        dqb[off] = __float2bfloat16(aq[j] * scale);
        dkb[off] = __float2bfloat16(ak[j] * scale);
        dvb[off] = __float2bfloat16(av[j]);
      }
    }
  }
}

}  // namespace

#ifndef DC_SAN_MAIN

at::Tensor banded_attn_train_bwd2(
    at::Tensor qkv, at::Tensor p, at::Tensor mask, at::Tensor dout,
    int64_t H, int64_t win, double p_drop) {
  auto qc = qkv.contiguous();
  auto pc = p.contiguous();
  auto dc = dout.contiguous();
  const int B = qc.size(0), T = qc.size(1);
  const int D = qc.size(2) / (3 * (int)H);
  TORCH_CHECK(T <= BT_MAXT && D <= BT_MAXD, "shape exceeds limits");
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
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(battn_bwd2_kernel, dim3(B * (int)H), dim3(256), 0,
                     stream, (const bf16_t*)qc.data_ptr(),
                     (const bf16_t*)pc.data_ptr(), mptr,
                     (const bf16_t*)dc.data_ptr(),
                     (bf16_t*)dqkv.data_ptr(), B * (int)H, (int)H, T, D,
                     (int)win, (float)(1.0 / std::sqrt((double)D)),
                     keep_inv);
  return dqkv;
}

#endif  // DC_SAN_MAIN
