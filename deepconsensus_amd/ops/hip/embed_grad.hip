// Fused embedding-stack backward (training) for gfx950.
//
// Computes the gradient of every embedding table of the learned-values
// model in ONE kernel. The autograd chain this replaces ran per-table
// sort-based scatters (sum_and_scatter + segment offsets + grad_weight)
// plus the permute/concat backward copies — together ~24% of the bf16
// train step (profiles/r01_train_top_kernels_postforeach.txt).
//
// Each workgroup accumulates its slab of (b, l) positions into an
// LDS-resident fp32 copy of ALL tables (~36 KB: vocab x width summed over
// unique tables), then merges once into the global gradient buffer with
// fp32 atomics — 256 atomic adds per table element total instead of one
// global scatter per id occurrence.
//
// Per input row r the host supplies: the source table's element offset
// (shared tables alias, e.g. the CCS row reuses the bases table), the
// embedding width and concat column base, the id shift (+1 for ccs_bq),
// the vocab clamp, and the sqrt(width) forward scale. Id 0 (post-shift)
// is the masked padding row: its gradient contribution is dropped, like
// ScaledEmbedding's zero-mask.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

namespace {

constexpr int MAX_TBL_ELEMS = 12288;  // 48 KB fp32 LDS accumulator

__global__ __launch_bounds__(256) void embed_grad_kernel(
    const float* __restrict__ rows,      // [B, R, L]
    const float* __restrict__ grad_out,  // [B, L, C]
    const int* __restrict__ row_shift,   // [R]
    const int* __restrict__ row_vocab,   // [R]
    const int* __restrict__ row_tbase,   // [R] table elem offset
    const int* __restrict__ row_width,   // [R]
    const int* __restrict__ row_col,     // [R] concat col base
    const float* __restrict__ row_scale, // [R] sqrt(width)
    float* __restrict__ grad_tables,     // [tbl_elems]
    int B, int R, int L, int C, int tbl_elems) {
  __shared__ float acc[MAX_TBL_ELEMS];

  const int tid = threadIdx.x;
  for (int i = tid; i < tbl_elems; i += 256) acc[i] = 0.f;
  __syncthreads();

  // Slab of positions for this block (grid-stride over B*L).
  const int npos = B * L;
  for (int pos = blockIdx.x * 256 + tid; pos < npos;
       pos += gridDim.x * 256) {
    const int b = pos / L, l = pos % L;
    const float* go = grad_out + ((size_t)b * L + l) * C;
    for (int r = 0; r < R; ++r) {
      int id = (int)rows[((size_t)b * R + r) * L + l] + row_shift[r];
      const int vmax = row_vocab[r] - 1;
      id = id < 0 ? 0 : (id > vmax ? vmax : id);
      if (id == 0) continue;  // masked padding row: no gradient
      const int w = row_width[r];
      const float scale = row_scale[r];
      const int base = row_tbase[r] + id * w;
      const int col = row_col[r];
      for (int j = 0; j < w; ++j) {
        atomicAdd(&acc[base + j], go[col + j] * scale);
      }
    }
  }
  __syncthreads();

  for (int i = tid; i < tbl_elems; i += 256) {
    const float v = acc[i];
    if (v != 0.f) atomicAdd(&grad_tables[i], v);
  }
}

}  // namespace

void embed_grad(at::Tensor rows, at::Tensor grad_out, at::Tensor row_shift,
                at::Tensor row_vocab, at::Tensor row_tbase,
                at::Tensor row_width, at::Tensor row_col,
                at::Tensor row_scale, at::Tensor grad_tables) {
  TORCH_CHECK(rows.is_cuda() && rows.dtype() == at::kFloat,
              "rows must be fp32 on device");
  TORCH_CHECK(grad_out.is_cuda() && grad_out.dtype() == at::kFloat,
              "grad_out must be fp32 on device");
  auto rc = rows.contiguous();
  auto gc = grad_out.contiguous();
  const int B = rc.size(0), R = rc.size(1), L = rc.size(2);
  const int C = gc.size(-1);
  const int tbl = grad_tables.numel();
  TORCH_CHECK(tbl <= MAX_TBL_ELEMS, "table too large for LDS accumulator");
  TORCH_CHECK(gc.numel() == (int64_t)B * L * C, "grad_out shape");
  dim3 grid(std::min((B * L + 255) / 256, 2048));
  dim3 block(256);
  hipStream_t stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(embed_grad_kernel, grid, block, 0, stream,
                     rc.data_ptr<float>(), gc.data_ptr<float>(),
                     row_shift.data_ptr<int>(), row_vocab.data_ptr<int>(),
                     row_tbase.data_ptr<int>(), row_width.data_ptr<int>(),
                     row_col.data_ptr<int>(), row_scale.data_ptr<float>(),
                     grad_tables.data_ptr<float>(), B, R, L, C, tbl);
}
