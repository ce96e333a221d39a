// DeepConsensus-AMD HIP/CDNA4 kernels (gfx950 / MI355X).
//
// Kernel inventory:
//  * embed_gather (embed_gather.hip) — K2; banded_attn (banded_attn.hip)
//    — K5-K7 core.
//  * fused_ln_head_qv — K10+K11+K12: final LayerNorm (eps 1e-6, fp32,
//    encoder_stack.py:131-133) + Dense(5) head (networks.py:342-345) +
//    softmax + argmax + Phred QV with linear calibration and cap
//    (quick_inference.py:377-389) fused into one wave-per-position kernel
//    emitting 2 bytes per position (base id + integer QV).
//
// Written CDNA4-first: 64-wide wavefronts, LDS staging for the id tile,
// fp32 accumulation, no CUDA-compat shims.

#ifndef DC_SAN_MAIN
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#endif
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DC_CHECK(x, msg) TORCH_CHECK(x, msg)

using bf16 = __hip_bfloat16;

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
  // Persistent waves: one-shot wave-per-position launched N/4 tiny blocks
  // and was launch/drain-bound (~11x the HBM time at N=409600).
  for (int n = blockIdx.x * 4 + wave_in_block; n < N; n += gridDim.x * 4) {
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
}

}  // namespace

#ifndef DC_SAN_MAIN

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
  dim3 grid(std::min((N + 3) / 4, 2048));
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

// Defined in banded_attn.hip / embed_gather.hip.
at::Tensor banded_attn(at::Tensor qkv, int64_t H, int64_t win);
at::Tensor banded_attn_mfma(at::Tensor qkv, int64_t H, int64_t win,
                            double scale);
at::Tensor embed_gather(at::Tensor rows, at::Tensor table_flat,
                        at::Tensor row_shift, at::Tensor row_vocab,
                        at::Tensor chunk_cnt, at::Tensor chunk_entries);
at::Tensor fused_ffn(at::Tensor x, at::Tensor w1, at::Tensor b1,
                     at::Tensor w2, at::Tensor b2, double alpha);
at::Tensor fused_ffn_v2(at::Tensor x, at::Tensor w1, at::Tensor w2,
                        at::Tensor b2, double alpha);
at::Tensor fused_ffn_v3(at::Tensor x, at::Tensor w1, at::Tensor w2,
                        at::Tensor b2, double alpha);
void embed_grad(at::Tensor rows, at::Tensor grad_out, at::Tensor row_shift,
                at::Tensor row_vocab, at::Tensor row_tbase,
                at::Tensor row_width, at::Tensor row_col,
                at::Tensor row_scale, at::Tensor grad_tables);
at::Tensor fused_linear(at::Tensor x, at::Tensor w, at::Tensor bias,
                        at::Tensor resid, int64_t n_out, bool relu,
                        double alpha);
std::vector<at::Tensor> alignment_dp_fwd(at::Tensor subs, at::Tensor ins,
                                         at::Tensor seq_lens, double del_cost,
                                         double reg, int64_t width);
std::vector<at::Tensor> alignment_dp_bwd(at::Tensor grad_out,
                                         at::Tensor weights,
                                         at::Tensor seq_lens, int64_t m,
                                         int64_t n, int64_t width);
std::vector<at::Tensor> alignment_metric_counts(
    at::Tensor yt, at::Tensor yp, at::Tensor yt_len, at::Tensor yp_len,
    double ms, double mp, double go, double ge);
at::Tensor ffn_ablate(at::Tensor x, at::Tensor w1, at::Tensor w2,
                      at::Tensor b2, double alpha, int64_t mode);
at::Tensor fused_ffn_v4(at::Tensor x, at::Tensor w1, at::Tensor w2,
                        at::Tensor b2, double alpha);
std::vector<at::Tensor> banded_attn_train_fwd(
    at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor mask,
    int64_t win, double p_drop);
std::vector<at::Tensor> banded_attn_train_bwd(
    at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor p,
    at::Tensor mask, at::Tensor dout, int64_t win, double p_drop);
std::vector<at::Tensor> banded_attn_mfma_train_fwd(
    at::Tensor qkv, int64_t H, int64_t win, double scale,
    at::Tensor drop_mask, double p_drop);
at::Tensor banded_attn_train_bwd2(
    at::Tensor qkv, at::Tensor p, at::Tensor mask, at::Tensor dout,
    int64_t H, int64_t win, double p_drop);
at::Tensor banded_attn_bwd_mfma(
    at::Tensor qkv, at::Tensor p, at::Tensor mask, at::Tensor dout,
    int64_t H, int64_t win, double p_drop);
at::Tensor fused_condense(at::Tensor x, at::Tensor w, at::Tensor pos,
                          int64_t n_out, int64_t seq_len);
std::vector<at::Tensor> ffn_train_fwd(at::Tensor x, at::Tensor w1,
                                      at::Tensor w2, at::Tensor b2,
                                      double p_drop, int64_t seed);
std::vector<at::Tensor> ffn_train_dgrad(at::Tensor dy, at::Tensor hd,
                                        at::Tensor w2t, at::Tensor w1t,
                                        double p_drop);
std::vector<at::Tensor> ffn_train_dgrad_nomask(at::Tensor dy, at::Tensor hd,
                                               at::Tensor w2t,
                                               at::Tensor w1t,
                                               double p_drop);
at::Tensor resid_drop_fwd(at::Tensor x, at::Tensor y, double alpha,
                          double p_drop, int64_t seed);
std::vector<at::Tensor> resid_drop_bwd(at::Tensor dout, at::Tensor y,
                                       double alpha, double p_drop,
                                       int64_t seed);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_condense", &fused_condense,
        "Condenser GEMM [M,560]x[560,N] + fused position-encoding add "
        "(K3+K4)");
  m.def("ffn_train_fwd", &ffn_train_fwd,
        "Training FFN forward: relu + hash dropout in-register, returns "
        "(y, hd)");
  m.def("ffn_train_dgrad", &ffn_train_dgrad,
        "Training FFN dgrad: both transposed GEMMs fused, mask from hd>0, "
        "returns (dx, dh_pre)");
  m.def("ffn_train_dgrad_nomask", &ffn_train_dgrad_nomask,
        "Bisection probe: dgrad without the mask (dh == dy@W2^T)");
  m.def("resid_drop_fwd", &resid_drop_fwd,
        "Fused ReZero residual + hash-dropout forward");
  m.def("resid_drop_bwd", &resid_drop_bwd,
        "Fused residual-dropout backward: (dy, dalpha partials)");
  m.def("fused_ln_head_qv", &fused_ln_head_qv,
        "Fused final LayerNorm + head + softmax + QV (K10+K11+K12)");
  m.def("banded_attn", &banded_attn,
        "Banded multi-head self-attention forward (K5-K7 core, generic)");
  m.def("banded_attn_mfma", &banded_attn_mfma,
        "Banded MHA forward on MFMA (D=140, L<=104)");
  m.def("embed_gather", &embed_gather,
        "Subread-stack embedding gather (K2)");
  m.def("fused_ffn", &fused_ffn,
        "Fused FFN + ReZero residual (K8+K9), hidden tensor LDS-resident");
  m.def("fused_ffn_v2", &fused_ffn_v2,
        "Fused FFN v2: glds-pipelined weights, bias folded into W1");
  m.def("fused_ffn_v3", &fused_ffn_v3,
        "Fused FFN v3: 256-row tiles, register-resident h (swapped B1)");
  m.def("embed_grad", &embed_grad,
        "Fused embedding-stack backward (training, all tables, one pass)");
  m.def("fused_linear", &fused_linear,
        "Fused linear projection (K5/K7): act(xW^T+b)[*alpha+x]");
  m.def("alignment_dp_fwd", &alignment_dp_fwd,
        "Alignment-loss wavefront DP forward (K13)");
  m.def("alignment_dp_bwd", &alignment_dp_bwd,
        "Alignment-loss wavefront DP backward (K13 VJP)");
  m.def("alignment_metric_counts", &alignment_metric_counts,
        "Device AlignmentMetric (K14): affine NW + backtrace counts");
  m.def("banded_attn_train_fwd", &banded_attn_train_fwd,
        "Training banded attention forward (saves band softmax P)");
  m.def("banded_attn_mfma_train_fwd", &banded_attn_mfma_train_fwd,
        "MFMA training attention forward (serving kernel + band-P save "
        "+ fused dropout)");
  m.def("banded_attn_bwd_mfma", &banded_attn_bwd_mfma,
        "MFMA training attention backward (band_scores/band_apply "
        "primitives; packed dqkv)");
  m.def("banded_attn_train_bwd2", &banded_attn_train_bwd2,
        "Training attention backward v2 (packed qkv layout, 2 blocks/"
        "CU) -> packed dqkv");
  m.def("banded_attn_train_bwd", &banded_attn_train_bwd,
        "Training banded attention backward (band-local dS -> dq,dk,dv)");
  m.def("fused_ffn_v4", &fused_ffn_v4,
        "Fused FFN v4: B1 on 16x16x32 MFMAs (4 independent chains)");
  m.def("ffn_ablate", &ffn_ablate,
        "fused_ffn_v3 ablation probe (0 full, 1 B1-only, 2 B2-only, "
        "3 loads-only) — perf diagnosis");
}

#endif  // DC_SAN_MAIN
