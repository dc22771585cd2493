// Rotary position embedding (RoPE, neox/llama rotate-half convention)
// forward + backward for CDNA4 (gfx950).
//
// cos/sin tables are precomputed on the HOST (guide: on-device trig turns
// a memory-bound op VALU-bound) as fp32 [max_pos, Dh/2].
//
//   y[p, h, i]        = x[i]  * cos[p, i] - x[i+Dh/2] * sin[p, i]
//   y[p, h, i+Dh/2]   = x[i+Dh/2] * cos[p, i] + x[i]  * sin[p, i]
//
// Backward is the inverse rotation (sin negated) — same kernel.
//
// x layout: [rows, Dh] contiguous, where row r belongs to token r / H
// (H heads per token) and positions[token] gives p. q and k are rotated in
// one call each from Python.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;

// Each thread handles VP=4 (i, i+Dh/2) pairs: 8-byte bf16x4 loads per half.
template <typename T, bool BWD>
__global__ void rope_kernel(const T *__restrict__ x, T *__restrict__ y,
                            const float *__restrict__ cs,   // [max_pos, half] cos
                            const float *__restrict__ sn,   // [max_pos, half] sin
                            const int32_t *__restrict__ pos,  // [n_tokens]
                            int64_t n_rows, int heads, int half) {
  constexpr int VP = 4;
  const int nvec = half / VP;
  const int64_t total = n_rows * nvec;
  const int64_t stride = (int64_t)gridDim.x * BLOCK;

  for (int64_t idx = (int64_t)blockIdx.x * BLOCK + threadIdx.x; idx < total;
       idx += stride) {
    const int64_t row = idx / nvec;
    const int v = (int)(idx % nvec);
    const int64_t token = row / heads;
    const int p = pos[token];

    const T *x1 = x + row * 2 * half + v * VP;
    const T *x2 = x1 + half;
    T *y1 = y + row * 2 * half + v * VP;
    T *y2 = y1 + half;
    const float *c = cs + (int64_t)p * half + v * VP;
    const float *s = sn + (int64_t)p * half + v * VP;

#pragma unroll
    for (int k = 0; k < VP; ++k) {
      const float a = rb::bf16_to_f32_or_id(x1[k]);
      const float b = rb::bf16_to_f32_or_id(x2[k]);
      const float sk = BWD ? -s[k] : s[k];
      rb::store_scalar(y1 + k, a * c[k] - b * sk);
      rb::store_scalar(y2 + k, b * c[k] + a * sk);
    }
  }
}

}  // namespace

template <bool BWD>
static at::Tensor rope_apply_impl(at::Tensor x, at::Tensor cos, at::Tensor sin,
                                  at::Tensor positions, int64_t heads) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "rope: x must be contiguous GPU");
  TORCH_CHECK(cos.scalar_type() == at::kFloat && sin.scalar_type() == at::kFloat,
              "rope: cos/sin must be fp32");
  TORCH_CHECK(positions.scalar_type() == at::kInt, "rope: positions must be int32");
  const int Dh = (int)x.size(-1);
  const int half = Dh / 2;
  TORCH_CHECK(half % 4 == 0, "rope: Dh/2 must be a multiple of 4");
  TORCH_CHECK(cos.size(-1) == half, "rope: cos table must be [max_pos, Dh/2]");
  const int64_t n_rows = x.numel() / Dh;
  auto y = at::empty_like(x);
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = rb::rb_grid_1d(n_rows * (half / 4), BLOCK);

  if (x.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL((rope_kernel<uint16_t, BWD>), dim3(grid), dim3(BLOCK), 0, stream,
                       (const uint16_t *)x.data_ptr(), (uint16_t *)y.data_ptr(),
                       cos.data_ptr<float>(), sin.data_ptr<float>(),
                       positions.data_ptr<int32_t>(), n_rows, (int)heads, half);
  } else if (x.scalar_type() == at::kFloat) {
    hipLaunchKernelGGL((rope_kernel<float, BWD>), dim3(grid), dim3(BLOCK), 0, stream,
                       x.data_ptr<float>(), y.data_ptr<float>(),
                       cos.data_ptr<float>(), sin.data_ptr<float>(),
                       positions.data_ptr<int32_t>(), n_rows, (int)heads, half);
  } else {
    TORCH_CHECK(false, "rope: unsupported dtype");
  }
  return y;
}

at::Tensor rope_fwd(at::Tensor x, at::Tensor cos, at::Tensor sin,
                    at::Tensor positions, int64_t heads) {
  return rope_apply_impl<false>(x, cos, sin, positions, heads);
}

at::Tensor rope_bwd(at::Tensor dy, at::Tensor cos, at::Tensor sin,
                    at::Tensor positions, int64_t heads) {
  return rope_apply_impl<true>(dy, cos, sin, positions, heads);
}
