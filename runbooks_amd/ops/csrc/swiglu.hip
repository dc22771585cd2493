// Fused SwiGLU activation for CDNA4 (gfx950).
//
//   fwd: y = silu(g) * u          (one pass vs eager's 3 kernels)
//   bwd: dg = dy * u * silu'(g);  du = dy * silu(g)   (one pass vs ~6)
//
// g, u are the gate/up GEMM outputs [T, I]; all math fp32, IO bf16/f32
// vectorized 16 B/lane. silu'(x) = s(x) * (1 + x * (1 - s(x))).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;

RB_DEV float sigmoidf(float x) { return 1.0f / (1.0f + __expf(-x)); }

template <typename T>
__global__ void swiglu_fwd_kernel(const T *__restrict__ g, const T *__restrict__ u,
                                  T *__restrict__ y, int64_t n) {
  constexpr int W = rb::VIO<T>::W;
  const int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x; i * W < n; i += stride) {
    float gf[W], uf[W];
    rb::VIO<T>::load(g + i * W, gf);
    rb::VIO<T>::load(u + i * W, uf);
#pragma unroll
    for (int e = 0; e < W; ++e) gf[e] = gf[e] * sigmoidf(gf[e]) * uf[e];
    rb::VIO<T>::store(y + i * W, gf);
  }
}

template <typename T>
__global__ void swiglu_bwd_kernel(const T *__restrict__ dy, const T *__restrict__ g,
                                  const T *__restrict__ u, T *__restrict__ dg,
                                  T *__restrict__ du, int64_t n) {
  constexpr int W = rb::VIO<T>::W;
  const int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x; i * W < n; i += stride) {
    float dyf[W], gf[W], uf[W], dgf[W];
    rb::VIO<T>::load(dy + i * W, dyf);
    rb::VIO<T>::load(g + i * W, gf);
    rb::VIO<T>::load(u + i * W, uf);
#pragma unroll
    for (int e = 0; e < W; ++e) {
      const float s = sigmoidf(gf[e]);
      const float silu = gf[e] * s;
      dgf[e] = dyf[e] * uf[e] * (s * (1.0f + gf[e] * (1.0f - s)));
      uf[e] = dyf[e] * silu;           // du
    }
    rb::VIO<T>::store(dg + i * W, dgf);
    rb::VIO<T>::store(du + i * W, uf);
  }
}

}  // namespace

at::Tensor swiglu_fwd(at::Tensor g, at::Tensor u) {
  TORCH_CHECK(g.is_cuda() && g.is_contiguous() && u.is_contiguous(), "swiglu: args");
  auto y = at::empty_like(g);
  const int64_t n = g.numel();
  auto stream = at::hip::getCurrentHIPStream();
  if (g.scalar_type() == at::kBFloat16) {
    TORCH_CHECK(n % 8 == 0, "swiglu bf16: numel % 8");
    const int grid = rb::rb_grid_1d(n / 8, BLOCK);
    hipLaunchKernelGGL(swiglu_fwd_kernel<uint16_t>, dim3(grid), dim3(BLOCK), 0, stream,
                       (const uint16_t *)g.data_ptr(), (const uint16_t *)u.data_ptr(),
                       (uint16_t *)y.data_ptr(), n);
  } else {
    TORCH_CHECK(g.scalar_type() == at::kFloat && n % 4 == 0, "swiglu dtype");
    const int grid = rb::rb_grid_1d(n / 4, BLOCK);
    hipLaunchKernelGGL(swiglu_fwd_kernel<float>, dim3(grid), dim3(BLOCK), 0, stream,
                       g.data_ptr<float>(), u.data_ptr<float>(), y.data_ptr<float>(), n);
  }
  return y;
}

std::vector<at::Tensor> swiglu_bwd(at::Tensor dy, at::Tensor g, at::Tensor u) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous(), "swiglu_bwd: args");
  auto dg = at::empty_like(g);
  auto du = at::empty_like(u);
  const int64_t n = g.numel();
  auto stream = at::hip::getCurrentHIPStream();
  if (g.scalar_type() == at::kBFloat16) {
    const int grid = rb::rb_grid_1d(n / 8, BLOCK);
    hipLaunchKernelGGL(swiglu_bwd_kernel<uint16_t>, dim3(grid), dim3(BLOCK), 0, stream,
                       (const uint16_t *)dy.data_ptr(), (const uint16_t *)g.data_ptr(),
                       (const uint16_t *)u.data_ptr(), (uint16_t *)dg.data_ptr(),
                       (uint16_t *)du.data_ptr(), n);
  } else {
    const int grid = rb::rb_grid_1d(n / 4, BLOCK);
    hipLaunchKernelGGL(swiglu_bwd_kernel<float>, dim3(grid), dim3(BLOCK), 0, stream,
                       dy.data_ptr<float>(), g.data_ptr<float>(), u.data_ptr<float>(),
                       dg.data_ptr<float>(), du.data_ptr<float>(), n);
  }
  return {dg, du};
}
