// Fused AdamW for CDNA4 (gfx950).
//
// One HBM pass per tensor per step: reads {param, grad, exp_avg,
// exp_avg_sq}, writes {param, exp_avg, exp_avg_sq} — vs the ~10
// elementwise kernels of eager torch AdamW. Decoupled weight decay
// (Loshchilov & Hutter), bias-corrected, matching
// torch.optim.AdamW(foreach=False) semantics.
//
// Params may be fp32 (master weights) or bf16 (LoRA adapters trained in
// low precision keep fp32 moments). Grad dtype may differ from param
// dtype (bf16 grads + fp32 master params).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;

template <typename PT, typename GT>
__global__ void adamw_kernel(PT *__restrict__ p, const GT *__restrict__ g,
                             float *__restrict__ m, float *__restrict__ v,
                             int64_t n, float lr, float beta1, float beta2,
                             float eps, float wd, float bc1, float bc2) {
  const int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x; i < n; i += stride) {
    const float gi = rb::bf16_to_f32_or_id(g[i]);
    float pi = rb::bf16_to_f32_or_id(p[i]);
    float mi = m[i];
    float vi = v[i];
    mi = beta1 * mi + (1.0f - beta1) * gi;
    vi = beta2 * vi + (1.0f - beta2) * gi * gi;
    const float mhat = mi / bc1;
    const float vhat = vi / bc2;
    pi -= lr * (mhat / (sqrtf(vhat) + eps) + wd * pi);
    m[i] = mi;
    v[i] = vi;
    rb::store_scalar(&p[i], pi);
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// Multi-tensor AdamW: one launch for the whole (dtype-uniform) param group.
// The chunk table is built once by the Python optimizer (pointers are
// stable: grads are DDP bucket views, moments allocated once) and replayed
// every step. Table row: {p, g, m, v, n} as int64.
// ---------------------------------------------------------------------------
namespace {

constexpr int MT_CHUNK = 1 << 16;  // elems per chunk = per block

template <typename PT, typename GT>
__global__ void adamw_mt_kernel(const int64_t *__restrict__ table, float lr,
                                float beta1, float beta2, float eps, float wd,
                                float bc1, float bc2) {
  const int64_t *row = table + (int64_t)blockIdx.x * 5;
  PT *p = (PT *)row[0];
  const GT *g = (const GT *)row[1];
  float *m = (float *)row[2];
  float *v = (float *)row[3];
  const int64_t n = row[4];
  for (int64_t i = threadIdx.x; i < n; i += BLOCK) {
    const float gi = rb::bf16_to_f32_or_id(g[i]);
    float pi = rb::bf16_to_f32_or_id(p[i]);
    float mi = m[i];
    float vi = v[i];
    mi = beta1 * mi + (1.0f - beta1) * gi;
    vi = beta2 * vi + (1.0f - beta2) * gi * gi;
    pi -= lr * ((mi / bc1) / (sqrtf(vi / bc2) + eps) + wd * pi);
    m[i] = mi;
    v[i] = vi;
    rb::store_scalar(&p[i], pi);
  }
}

}  // namespace

int64_t adamw_mt_chunk_elems() { return MT_CHUNK; }

void adamw_step_multi(at::Tensor table, int64_t nchunks, bool p32, bool g32,
                      double lr, double beta1, double beta2, double eps,
                      double wd, int64_t step) {
  TORCH_CHECK(table.is_cuda() && table.scalar_type() == at::kLong &&
                  table.is_contiguous(), "adamw_mt: bad table");
  const float bc1 = 1.0f - powf((float)beta1, (float)step);
  const float bc2 = 1.0f - powf((float)beta2, (float)step);
  auto stream = at::hip::getCurrentHIPStream();
#define RB_ADAMW_MT(PT, GT)                                                   \
  hipLaunchKernelGGL((adamw_mt_kernel<PT, GT>), dim3((uint32_t)nchunks),      \
                     dim3(BLOCK), 0, stream, table.data_ptr<int64_t>(),       \
                     (float)lr, (float)beta1, (float)beta2, (float)eps,       \
                     (float)wd, bc1, bc2)
  if (p32 && g32) RB_ADAMW_MT(float, float);
  else if (p32 && !g32) RB_ADAMW_MT(float, uint16_t);
  else if (!p32 && g32) RB_ADAMW_MT(uint16_t, float);
  else RB_ADAMW_MT(uint16_t, uint16_t);
#undef RB_ADAMW_MT
}

void adamw_step(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v,
                double lr, double beta1, double beta2, double eps, double wd,
                int64_t step) {
  TORCH_CHECK(p.is_cuda() && p.is_contiguous() && g.is_contiguous() &&
                  m.is_contiguous() && v.is_contiguous(),
              "adamw: contiguous GPU tensors required");
  TORCH_CHECK(m.scalar_type() == at::kFloat && v.scalar_type() == at::kFloat,
              "adamw: moments must be fp32");
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == m.numel(), "adamw: size mismatch");
  const int64_t n = p.numel();
  const float bc1 = 1.0f - powf((float)beta1, (float)step);
  const float bc2 = 1.0f - powf((float)beta2, (float)step);
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = rb::rb_grid_1d(n, BLOCK);

#define RB_ADAMW_LAUNCH(PT, GT)                                                   \
  hipLaunchKernelGGL((adamw_kernel<PT, GT>), dim3(grid), dim3(BLOCK), 0, stream,  \
                     (PT *)p.data_ptr(), (const GT *)g.data_ptr(),                \
                     m.data_ptr<float>(), v.data_ptr<float>(), n, (float)lr,      \
                     (float)beta1, (float)beta2, (float)eps, (float)wd, bc1, bc2)

  const bool p32 = p.scalar_type() == at::kFloat;
  const bool g32 = g.scalar_type() == at::kFloat;
  TORCH_CHECK(p32 || p.scalar_type() == at::kBFloat16, "adamw: param dtype");
  TORCH_CHECK(g32 || g.scalar_type() == at::kBFloat16, "adamw: grad dtype");
  if (p32 && g32) RB_ADAMW_LAUNCH(float, float);
  else if (p32 && !g32) RB_ADAMW_LAUNCH(float, uint16_t);
  else if (!p32 && g32) RB_ADAMW_LAUNCH(uint16_t, float);
  else RB_ADAMW_LAUNCH(uint16_t, uint16_t);
#undef RB_ADAMW_LAUNCH
}
