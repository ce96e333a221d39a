// Device AlignmentMetric (K14) for gfx950: PBMM2-approximating 3-state
// affine-gap Needleman-Wunsch + backtrace counts.
//
// Reference semantics: losses_and_metrics.py:666-1058 (and the numpy
// oracle in models/losses.py AlignmentMetric.alignment): states
// {M, I, D}, scores match +ms / mismatch -mp, gap open -go / extend -ge
// with PBMM2's open = open + extend convention folded in host-side;
// forward runs anti-diagonal wavefronts with per-state argmax direction
// tracking (ties first-max: M, I, D order), then a per-example
// backtrace classifies each edge by its source state and counts
// matches / insertions / deletions / correct matches — all four feed
// pid and the yield metric without leaving the device.
//
// Forward: one 128-thread block per example, diagonals in LDS,
// direction tensor [B][m+n+1][3][m+1] int8 in global.
// Backtrace: one lane per example (divergent walk, <= m+n steps).

#ifndef DC_SAN_MAIN
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#endif
#include <hip/hip_runtime.h>

namespace {

constexpr int MAXW = 160;  // max (m+1)
constexpr float NEG = -1e30f;

__global__ void metric_fwd_kernel(
    const int* __restrict__ yt,      // [B, m] left-shifted tokens
    const int* __restrict__ yp,      // [B, n]
    const int* __restrict__ yt_len,  // [B]
    const int* __restrict__ yp_len,  // [B]
    float* __restrict__ v_opt,       // [B]
    int* __restrict__ m_opt,         // [B]
    int8_t* __restrict__ dir_all,    // [B, m+n+1, 3, m+1]
    int B, int m, int n,
    float ms, float mp, float go, float ge) {
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  __shared__ float vbuf[3][3][MAXW];  // [diag age 2,1,cur][state][i]
  float (*vm2)[MAXW] = vbuf[0];
  float (*vm1)[MAXW] = vbuf[1];
  float (*vcur)[MAXW] = vbuf[2];
  __shared__ float s_vopt;
  __shared__ int s_mopt;

  const int* ytb = yt + (size_t)b * m;
  const int* ypb = yp + (size_t)b * n;
  const int sl = yt_len[b];
  const int pl = yp_len[b];
  const int kend = sl + pl;
  int8_t* db = dir_all + (size_t)b * (m + n + 1) * 3 * (m + 1);

  for (int i = tid; i < MAXW; i += blockDim.x) {
    for (int s = 0; s < 3; ++s) {
      vm2[s][i] = NEG;
      vm1[s][i] = NEG;
      vcur[s][i] = NEG;
    }
  }
  // dir default -2 handled host-side (tensor filled with -2).
  if (tid == 0) {
    s_vopt = 0.f;
    s_mopt = -1;
    // k=0: V_M(0,0) = 0, dir -1 (start sentinel).
    vm2[0][0] = 0.f;
    db[0 * 3 * (m + 1) + 0 * (m + 1) + 0] = -1;
    // k=1: V_I(0,1) = -go (dir 0), V_D(1,0) = -go (dir 0).
    vm1[1][0] = -go;
    vm1[2][1] = -go;
    db[1 * 3 * (m + 1) + 1 * (m + 1) + 0] = 0;
    db[1 * 3 * (m + 1) + 2 * (m + 1) + 1] = 0;
    // maybe_update(k=1): argmax over states at i = sl on diag 1.
    if (kend == 1) {
      float best = vm1[0][sl];
      int bs = 0;
      if (vm1[1][sl] > best) { best = vm1[1][sl]; bs = 1; }
      if (vm1[2][sl] > best) { best = vm1[2][sl]; bs = 2; }
      s_vopt = best;
      s_mopt = bs;
    }
  }
  __syncthreads();

  for (int k = 2; k <= m + n; ++k) {
    int8_t* dk = db + (size_t)k * 3 * (m + 1);
    for (int i = tid; i <= min(m, k); i += blockDim.x) {
      const int j = k - i;
      float vM = NEG, vI = NEG, vD = NEG;
      int dM = -2, dI = -2, dD = -2;
      if (j >= 0 && j <= n) {
        if (i >= 1 && j >= 1 && i <= m && j <= n) {
          const float sub =
              (ytb[i - 1] == ypb[j - 1]) ? ms : -mp;
          // match: predecessor any state at (i-1, j-1) = vm2[i-1].
          float best = vm2[0][i - 1];
          int bs = 0;
          if (vm2[1][i - 1] > best) { best = vm2[1][i - 1]; bs = 1; }
          if (vm2[2][i - 1] > best) { best = vm2[2][i - 1]; bs = 2; }
          if (best > NEG * 0.5f) {
            vM = best + sub;
            dM = bs;
          }
        }
        if (j >= 1) {
          // insertion: from M (-go) or I (-ge) at (i, j-1) = vm1[i].
          const float cm = vm1[0][i] - go;
          const float ci = vm1[1][i] - ge;
          vI = cm;
          dI = 0;
          if (ci > vI) { vI = ci; dI = 1; }
          if (vI < NEG * 0.5f) { vI = NEG; dI = -2; }
        }
        if (i >= 1) {
          // deletion: from M (-go), I (-go) or D (-ge) at (i-1, j).
          const float cm = vm1[0][i - 1] - go;
          const float ci = vm1[1][i - 1] - go;
          const float cd = vm1[2][i - 1] - ge;
          vD = cm;
          dD = 0;
          if (ci > vD) { vD = ci; dD = 1; }
          if (cd > vD) { vD = cd; dD = 2; }
          if (vD < NEG * 0.5f) { vD = NEG; dD = -2; }
        }
        dk[0 * (m + 1) + i] = (int8_t)dM;
        dk[1 * (m + 1) + i] = (int8_t)dI;
        dk[2 * (m + 1) + i] = (int8_t)dD;
      }
      vcur[0][i] = vM;
      vcur[1][i] = vI;
      vcur[2][i] = vD;
      if (k == kend && i == sl) {
        float best = vM;
        int bs = 0;
        if (vI > best) { best = vI; bs = 1; }
        if (vD > best) { best = vD; bs = 2; }
        s_vopt = best;
        s_mopt = bs;
      }
    }
    __syncthreads();
    float (*t)[MAXW] = vm2;
    vm2 = vm1;
    vm1 = vcur;
    vcur = t;
    for (int i = tid; i <= min(m, k + 1); i += blockDim.x) {
      vcur[0][i] = NEG;
      vcur[1][i] = NEG;
      vcur[2][i] = NEG;
    }
    __syncthreads();
  }
  if (tid == 0) {
    v_opt[b] = s_vopt;
    m_opt[b] = s_mopt;
  }
}

__global__ void metric_backtrace_kernel(
    const int* __restrict__ yt, const int* __restrict__ yp,
    const int* __restrict__ yt_len, const int* __restrict__ yp_len,
    const int* __restrict__ m_opt,
    const int8_t* __restrict__ dir_all,
    int* __restrict__ counts,  // [B, 4]: matches, ins, del, correct
    int B, int m, int n) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  const int8_t* db = dir_all + (size_t)b * (m + n + 1) * 3 * (m + 1);
  const int* ytb = yt + (size_t)b * m;
  const int* ypb = yp + (size_t)b * n;
  const int sl = yt_len[b];
  const int pl = yp_len[b];
  int k = sl + pl;
  int i = sl;
  int mc = m_opt[b];
  int matches = 0, ins = 0, dels = 0, correct = 0;
  // steps per source state: match consumes (i-1, j-1), ins (j-1),
  // del (i-1).
  for (int step = 0; step <= m + n; ++step) {
    if (mc < 0 || k < 0 || i < 0 || i > m) break;
    const int j = k - i;
    if (j < 0 || j > n) break;
    const int mn = db[(size_t)k * 3 * (m + 1) + mc * (m + 1) + i];
    if (mn == -1) break;  // reached the (0,0) start sentinel
    if (mn == -2) break;  // malformed (defensive)
    if (mc == 0) {
      ++matches;
      if (i >= 1 && j >= 1 && ytb[i - 1] == ypb[j - 1]) ++correct;
      k -= 2;
      i -= 1;
    } else if (mc == 1) {
      ++ins;
      k -= 1;
    } else {
      ++dels;
      k -= 1;
      i -= 1;
    }
    mc = mn;
  }
  counts[(size_t)b * 4 + 0] = matches;
  counts[(size_t)b * 4 + 1] = ins;
  counts[(size_t)b * 4 + 2] = dels;
  counts[(size_t)b * 4 + 3] = correct;
}

}  // namespace

#ifndef DC_SAN_MAIN

std::vector<at::Tensor> alignment_metric_counts(
    at::Tensor yt, at::Tensor yp, at::Tensor yt_len, at::Tensor yp_len,
    double ms, double mp, double go, double ge) {
  TORCH_CHECK(yt.is_cuda() && yt.dtype() == at::kInt, "yt int32 cuda");
  TORCH_CHECK(yp.is_cuda() && yp.dtype() == at::kInt, "yp int32 cuda");
  auto ytc = yt.contiguous();
  auto ypc = yp.contiguous();
  auto ytl = yt_len.to(at::kInt).contiguous();
  auto ypl = yp_len.to(at::kInt).contiguous();
  const int B = ytc.size(0), m = ytc.size(1), n = ypc.size(1);
  TORCH_CHECK(m + 1 < MAXW, "window too long for metric kernel");
  auto v_opt = at::zeros({B}, ytc.options().dtype(at::kFloat));
  auto m_opt = at::full({B}, -1, ytc.options());
  auto dir_all = at::full({B, m + n + 1, 3, m + 1}, -2,
                          ytc.options().dtype(at::kChar));
  auto counts = at::zeros({B, 4}, ytc.options());
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(metric_fwd_kernel, dim3(B), dim3(128), 0, stream,
                     ytc.data_ptr<int>(), ypc.data_ptr<int>(),
                     ytl.data_ptr<int>(), ypl.data_ptr<int>(),
                     v_opt.data_ptr<float>(), m_opt.data_ptr<int>(),
                     dir_all.data_ptr<int8_t>(), B, m, n,
                     (float)ms, (float)mp, (float)go, (float)ge);
  hipLaunchKernelGGL(metric_backtrace_kernel,
                     dim3((B + 127) / 128), dim3(128), 0, stream,
                     ytc.data_ptr<int>(), ypc.data_ptr<int>(),
                     ytl.data_ptr<int>(), ypl.data_ptr<int>(),
                     m_opt.data_ptr<int>(), dir_all.data_ptr<int8_t>(),
                     counts.data_ptr<int>(), B, m, n);
  return {v_opt, counts};
}

#endif  // DC_SAN_MAIN
