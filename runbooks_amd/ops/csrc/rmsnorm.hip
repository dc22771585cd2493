// Fused RMSNorm forward/backward for CDNA4 (gfx950).
//
// Replaces the eager 4-kernel torch sequence (pow/mean/rsqrt/mul) with one
// HBM-bound pass per direction. Reference semantics match the plain
// PyTorch fp32 RMSNorm used by the test-suite reference models
// (llama-family norm; the reference platform delegates this to external
// HuggingFace trainer images — see SURVEY.md §2b).
//
// Layout: x [N, D] row-major. One workgroup (256 threads) per row,
// grid-strided. All math in fp32; bf16 IO vectorized 8-wide (16 B/lane).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;

// swz (optional, bf16 only): ALSO write the normed row into the decode
// GEMM's B-fragment lane order ([D/16][2][32][8], csrc/decode_gemm.hip)
// so the decode path needs no separate x-swizzle launch. Each thread's
// 8-element vector IS one swizzled 16 B slot: pos = (k/16)*512 +
// ((k/8)&1)*256 + row*8.
// res/sum_out (optional, together): normalize x + res instead of x and
// also write the sum — fuses the decode loop's residual add into the
// norm (one launch instead of two; the sum feeds the NEXT residual).
// res32/rsplit (optional, bf16 path): the residual arrives as the
// UNCOMBINED fp32 split-K slab [rsplit, n_rows, D] straight from
// decode_gemm_raw — folding the slices here saves the combine launch.
template <typename T>
__global__ void rmsnorm_fwd_kernel(const T *__restrict__ x,
                                   const T *__restrict__ res,
                                   const float *__restrict__ res32,
                                   int rsplit,
                                   T *__restrict__ sum_out,
                                   const T *__restrict__ w,
                                   T *__restrict__ y,
                                   float *__restrict__ inv_rms,
                                   uint16_t *__restrict__ swz,
                                   int64_t n_rows, int D, float eps) {
  constexpr int W = rb::VIO<T>::W;
  __shared__ float red[BLOCK / RB_WAVE];
  const int nvec = D / W;

  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const T *xr = x + row * D;
    T *yr = y + row * D;

    float ss = 0.0f;
    for (int i = threadIdx.x; i < nvec; i += BLOCK) {
      float f[W];
      rb::VIO<T>::load(xr + i * W, f);
      if (res != nullptr || res32 != nullptr) {
        if (res != nullptr) {
          float r[W];
          rb::VIO<T>::load(res + row * D + i * W, r);
#pragma unroll
          for (int k = 0; k < W; ++k) f[k] += r[k];
        } else {
          for (int sl = 0; sl < rsplit; ++sl) {
            const float *rp = res32 + ((int64_t)sl * n_rows + row) * D + i * W;
#pragma unroll
            for (int k = 0; k < W; k += 4) {
              float4 a = *reinterpret_cast<const float4 *>(rp + k);
              f[k] += a.x; f[k + 1] += a.y; f[k + 2] += a.z; f[k + 3] += a.w;
            }
          }
        }
        if (sum_out != nullptr)
          rb::VIO<T>::store(sum_out + row * D + i * W, f);
      }
#pragma unroll
      for (int k = 0; k < W; ++k) ss += f[k] * f[k];
    }
    ss = rb::block_reduce_sum(ss, red);
    const float ir = rsqrtf(ss / (float)D + eps);
    if (threadIdx.x == 0 && inv_rms != nullptr) inv_rms[row] = ir;

    for (int i = threadIdx.x; i < nvec; i += BLOCK) {
      float f[W], g[W];
      rb::VIO<T>::load(xr + i * W, f);     // L1-resident second read
      if (res != nullptr) {
        float r[W];
        rb::VIO<T>::load(res + row * D + i * W, r);
#pragma unroll
        for (int k = 0; k < W; ++k) f[k] += r[k];
      } else if (res32 != nullptr) {
        for (int sl = 0; sl < rsplit; ++sl) {
          const float *rp = res32 + ((int64_t)sl * n_rows + row) * D + i * W;
#pragma unroll
          for (int k = 0; k < W; k += 4) {
            float4 a = *reinterpret_cast<const float4 *>(rp + k);
            f[k] += a.x; f[k + 1] += a.y; f[k + 2] += a.z; f[k + 3] += a.w;
          }
        }
      }
      rb::VIO<T>::load(w + i * W, g);
#pragma unroll
      for (int k = 0; k < W; ++k) f[k] = f[k] * ir * g[k];
      rb::VIO<T>::store(yr + i * W, f);
      if constexpr (sizeof(T) == 2) {
        if (swz != nullptr) {
          const int k0 = i * 8;
          rb::VIO<T>::store(reinterpret_cast<T *>(swz) +
                            (k0 >> 4) * 512 + ((k0 >> 3) & 1) * 256 +
                            row * 8, f);
        }
      }
    }
    __syncthreads();
  }
}

// Backward: one workgroup per row (grid sized to give every row its own
// WG), dx in one pass. dw: each thread's column set is FIXED across its
// rows (i strides by BLOCK), so the per-thread dw partials live in
// REGISTERS (<= RB_RN_MAXV vectors = 32 floats at D=8192) and hit
// global memory once via atomicAdd at the end. The round-1 version
// accumulated dw element-wise in LDS — 16 scalar ds ops per 16 B of
// input made LDS the bottleneck (65 us on [2048,4096] vs ~14 ideal).
// D > 8*BLOCK*RB_RN_MAXV falls back to the LDS path.
// Register-dw variant, D == NV * BLOCK * W exactly (llama 4096 -> NV=2
// bf16). NV is a template int so every load is UNGUARDED — a runtime
// `if (i < nvec)` around unrolled loads makes hipcc branch per element
// and drain vmcnt(0) each time (cdna_hip_programming.md §5 trap (c));
// the first cut of this kernel did exactly that and ran 7x slower than
// the LDS version it meant to replace (r11: 472 us vs 65).
template <typename T, int NV>
__global__ void rmsnorm_bwd_reg_kernel(const T *__restrict__ x,
                                       const T *__restrict__ w,
                                       const T *__restrict__ dy,
                                       const float *__restrict__ inv_rms,
                                       T *__restrict__ dx,
                                       float *__restrict__ dw,
                                       int64_t n_rows, int D) {
  constexpr int W = rb::VIO<T>::W;
  __shared__ float red[BLOCK / RB_WAVE];

  float dwacc[NV][W];
#pragma unroll
  for (int v = 0; v < NV; ++v)
#pragma unroll
    for (int k = 0; k < W; ++k) dwacc[v][k] = 0.0f;

  float wf[NV][W];
#pragma unroll
  for (int v = 0; v < NV; ++v)
    rb::VIO<T>::load(w + (v * BLOCK + threadIdx.x) * W, wf[v]);

  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const T *xr = x + row * D;
    const T *dyr = dy + row * D;
    T *dxr = dx + row * D;
    const float ir = inv_rms[row];

    float xf[NV][W], df[NV][W];
    float s = 0.0f;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      const int i = v * BLOCK + threadIdx.x;
      rb::VIO<T>::load(xr + i * W, xf[v]);
      rb::VIO<T>::load(dyr + i * W, df[v]);
#pragma unroll
      for (int k = 0; k < W; ++k) s += df[v][k] * wf[v][k] * xf[v][k];
    }
    s = rb::block_reduce_sum(s, red);
    const float c = ir * ir * ir * s / (float)D;

#pragma unroll
    for (int v = 0; v < NV; ++v) {
      const int i = v * BLOCK + threadIdx.x;
      float o[W];
#pragma unroll
      for (int k = 0; k < W; ++k) {
        o[k] = ir * wf[v][k] * df[v][k] - c * xf[v][k];
        dwacc[v][k] += df[v][k] * xf[v][k] * ir;
      }
      rb::VIO<T>::store(dxr + i * W, o);
    }
  }

#pragma unroll
  for (int v = 0; v < NV; ++v) {
    const int i = v * BLOCK + threadIdx.x;
#pragma unroll
    for (int k = 0; k < W; ++k)
      if (dwacc[v][k] != 0.0f) atomicAdd(dw + i * W + k, dwacc[v][k]);
  }
}

template <typename T>
__global__ void rmsnorm_bwd_kernel(const T *__restrict__ x,
                                           const T *__restrict__ w,
                                           const T *__restrict__ dy,
                                           const float *__restrict__ inv_rms,
                                           T *__restrict__ dx,
                                           float *__restrict__ dw,
                                           int64_t n_rows, int D) {
  constexpr int W = rb::VIO<T>::W;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float *dw_loc = reinterpret_cast<float *>(smem);          // D floats
  float *red = dw_loc + D;                                  // BLOCK/64 floats
  const int nvec = D / W;

  for (int i = threadIdx.x; i < D; i += BLOCK) dw_loc[i] = 0.0f;
  __syncthreads();

  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const T *xr = x + row * D;
    const T *dyr = dy + row * D;
    T *dxr = dx + row * D;
    const float ir = inv_rms[row];

    float s = 0.0f;
    for (int i = threadIdx.x; i < nvec; i += BLOCK) {
      float xf[W], wf[W], df[W];
      rb::VIO<T>::load(xr + i * W, xf);
      rb::VIO<T>::load(w + i * W, wf);
      rb::VIO<T>::load(dyr + i * W, df);
#pragma unroll
      for (int k = 0; k < W; ++k) s += df[k] * wf[k] * xf[k];
    }
    s = rb::block_reduce_sum(s, red);
    const float c = ir * ir * ir * s / (float)D;

    for (int i = threadIdx.x; i < nvec; i += BLOCK) {
      float xf[W], wf[W], df[W], o[W];
      rb::VIO<T>::load(xr + i * W, xf);
      rb::VIO<T>::load(w + i * W, wf);
      rb::VIO<T>::load(dyr + i * W, df);
#pragma unroll
      for (int k = 0; k < W; ++k) {
        o[k] = ir * wf[k] * df[k] - c * xf[k];
        dw_loc[i * W + k] += df[k] * xf[k] * ir;
      }
      rb::VIO<T>::store(dxr + i * W, o);
    }
    __syncthreads();
  }

  __syncthreads();
  for (int i = threadIdx.x; i < D; i += BLOCK)
    if (dw_loc[i] != 0.0f) atomicAdd(dw + i, dw_loc[i]);
}

}  // namespace

std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && w.is_cuda(), "rmsnorm: tensors must be on GPU");
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous(), "rmsnorm: contiguous only");
  const int D = (int)x.size(-1);
  const int64_t n_rows = x.numel() / D;
  auto y = at::empty_like(x);
  auto inv_rms = at::empty({n_rows}, x.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();

  if (x.scalar_type() == at::kBFloat16) {
    TORCH_CHECK(D % 8 == 0, "rmsnorm bf16: D must be a multiple of 8");
    hipLaunchKernelGGL(rmsnorm_fwd_kernel<uint16_t>, dim3(std::min<int64_t>(n_rows, 2048)),
                       dim3(BLOCK), 0, stream,
                       (const uint16_t *)x.data_ptr(),
                       (const uint16_t *)nullptr, (const float *)nullptr, 0,
                       (uint16_t *)nullptr,
                       (const uint16_t *)w.data_ptr(),
                       (uint16_t *)y.data_ptr(), inv_rms.data_ptr<float>(),
                       (uint16_t *)nullptr, n_rows, D, (float)eps);
  } else if (x.scalar_type() == at::kFloat) {
    TORCH_CHECK(D % 4 == 0, "rmsnorm f32: D must be a multiple of 4");
    hipLaunchKernelGGL(rmsnorm_fwd_kernel<float>, dim3(std::min<int64_t>(n_rows, 2048)),
                       dim3(BLOCK), 0, stream,
                       x.data_ptr<float>(), (const float *)nullptr,
                       (const float *)nullptr, 0,
                       (float *)nullptr, w.data_ptr<float>(),
                       y.data_ptr<float>(),
                       inv_rms.data_ptr<float>(), (uint16_t *)nullptr,
                       n_rows, D, (float)eps);
  } else {
    TORCH_CHECK(false, "rmsnorm: unsupported dtype");
  }
  return {y, inv_rms};
}

// Decode-path variant: returns {y, y_swz} where y_swz is the decode
// GEMM's pre-swizzled B-operand ([D/16]x512 bf16, rows m >= n_rows left
// uninitialized: decode_gemm drops their outputs at the epilogue).
std::vector<at::Tensor> rmsnorm_fwd_dec(at::Tensor x, at::Tensor w,
                                        double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous(),
              "rmsnorm_fwd_dec: contiguous GPU tensors");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "rmsnorm_fwd_dec: bf16");
  const int64_t n_rows = x.numel() / x.size(-1);
  const int D = (int)x.size(-1);
  TORCH_CHECK(n_rows <= 32 && D % 16 == 0, "rmsnorm_fwd_dec: shape");
  auto y = at::empty_like(x);
  auto swz = at::empty({(int64_t)(D / 16) * 512}, x.options());
  auto stream = at::cuda::getCurrentHIPStream();
  auto inv_rms = at::empty({n_rows}, x.options().dtype(at::kFloat));
  hipLaunchKernelGGL(rmsnorm_fwd_kernel<uint16_t>, dim3(n_rows),
                     dim3(BLOCK), 0, stream,
                     (const uint16_t *)x.data_ptr(),
                     (const uint16_t *)nullptr, (const float *)nullptr, 0,
                     (uint16_t *)nullptr,
                     (const uint16_t *)w.data_ptr(),
                     (uint16_t *)y.data_ptr(), inv_rms.data_ptr<float>(),
                     (uint16_t *)swz.data_ptr(), n_rows, D, (float)eps);
  return {y, swz};
}

// Fused residual + norm + swizzle for the decode loop: returns
// {sum = x + res, y = rmsnorm(sum) * w, y_swz}. One launch replaces the
// eager residual add + the norm + the operand swizzle.
std::vector<at::Tensor> rmsnorm_res_fwd_dec(at::Tensor x, at::Tensor res,
                                            at::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && res.is_contiguous() &&
              w.is_contiguous(), "rmsnorm_res_fwd_dec: contiguous");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              res.scalar_type() == at::kBFloat16, "rmsnorm_res_fwd_dec: bf16");
  TORCH_CHECK(x.sizes() == res.sizes(), "rmsnorm_res_fwd_dec: shape");
  const int64_t n_rows = x.numel() / x.size(-1);
  const int D = (int)x.size(-1);
  TORCH_CHECK(n_rows <= 32 && D % 16 == 0, "rmsnorm_res_fwd_dec: shape");
  auto sum = at::empty_like(x);
  auto y = at::empty_like(x);
  auto swz = at::empty({(int64_t)(D / 16) * 512}, x.options());
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(rmsnorm_fwd_kernel<uint16_t>, dim3(n_rows),
                     dim3(BLOCK), 0, stream,
                     (const uint16_t *)x.data_ptr(),
                     (const uint16_t *)res.data_ptr(),
                     (const float *)nullptr, 0,
                     (uint16_t *)sum.data_ptr(),
                     (const uint16_t *)w.data_ptr(),
                     (uint16_t *)y.data_ptr(), (float *)nullptr,
                     (uint16_t *)swz.data_ptr(), n_rows, D, (float)eps);
  return {sum, y, swz};
}

// Same, with the residual as an UNCOMBINED fp32 split-K slab
// [split, n_rows, D] from decode_gemm_raw.
std::vector<at::Tensor> rmsnorm_res_slab_fwd_dec(at::Tensor x,
                                                 at::Tensor slabs,
                                                 at::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && slabs.is_contiguous() &&
              w.is_contiguous(), "rmsnorm_res_slab_fwd_dec: contiguous");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              slabs.scalar_type() == at::kFloat &&
              slabs.dim() == 3, "rmsnorm_res_slab_fwd_dec: dtypes");
  const int64_t n_rows = x.numel() / x.size(-1);
  const int D = (int)x.size(-1);
  TORCH_CHECK(slabs.size(1) == n_rows && slabs.size(2) == D &&
              n_rows <= 32 && D % 16 == 0, "rmsnorm_res_slab_fwd_dec: shape");
  auto sum = at::empty_like(x);
  auto y = at::empty_like(x);
  auto swz = at::empty({(int64_t)(D / 16) * 512}, x.options());
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(rmsnorm_fwd_kernel<uint16_t>, dim3(n_rows),
                     dim3(BLOCK), 0, stream,
                     (const uint16_t *)x.data_ptr(),
                     (const uint16_t *)nullptr,
                     (const float *)slabs.data_ptr(), (int)slabs.size(0),
                     (uint16_t *)sum.data_ptr(),
                     (const uint16_t *)w.data_ptr(),
                     (uint16_t *)y.data_ptr(), (float *)nullptr,
                     (uint16_t *)swz.data_ptr(), n_rows, D, (float)eps);
  return {sum, y, swz};
}

std::vector<at::Tensor> rmsnorm_bwd(at::Tensor x, at::Tensor w, at::Tensor dy,
                                    at::Tensor inv_rms) {
  TORCH_CHECK(x.is_cuda() && dy.is_contiguous() && x.is_contiguous(), "rmsnorm_bwd: bad args");
  const int D = (int)x.size(-1);
  const int64_t n_rows = x.numel() / D;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({D}, x.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  const int nwg = (int)std::min<int64_t>(n_rows, 4096);
  const size_t shmem = (size_t)D * sizeof(float) + (BLOCK / RB_WAVE) * sizeof(float);
  TORCH_CHECK(shmem <= 160 * 1024, "rmsnorm_bwd: D too large for LDS accumulation");

  // The register-dw variant measured 7x SLOWER than this LDS path on
  // [2048,4096] (472 vs 65 us, r11/r13) despite clean ISA (no spills,
  // 95 VGPR, 5 waves/SIMD) — root cause unidentified; the LDS
  // accumulate stays the shipped path.
  const int nv = 0;
  if (x.scalar_type() == at::kBFloat16) {
    auto a = (const uint16_t *)x.data_ptr();
    auto b = (const uint16_t *)w.data_ptr();
    auto c = (const uint16_t *)dy.data_ptr();
    auto irp = inv_rms.data_ptr<float>();
    auto dxp = (uint16_t *)dx.data_ptr();
    auto dwp = dw.data_ptr<float>();
    if (nv == 1)
      hipLaunchKernelGGL((rmsnorm_bwd_reg_kernel<uint16_t, 1>), dim3(nwg),
                         dim3(BLOCK), 0, stream, a, b, c, irp, dxp, dwp, n_rows, D);
    else if (nv == 2)
      hipLaunchKernelGGL((rmsnorm_bwd_reg_kernel<uint16_t, 2>), dim3(nwg),
                         dim3(BLOCK), 0, stream, a, b, c, irp, dxp, dwp, n_rows, D);
    else if (nv == 4)
      hipLaunchKernelGGL((rmsnorm_bwd_reg_kernel<uint16_t, 4>), dim3(nwg),
                         dim3(BLOCK), 0, stream, a, b, c, irp, dxp, dwp, n_rows, D);
    else
      hipLaunchKernelGGL(rmsnorm_bwd_kernel<uint16_t>, dim3(nwg), dim3(BLOCK),
                         shmem, stream, a, b, c, irp, dxp, dwp, n_rows, D);
  } else if (x.scalar_type() == at::kFloat) {
    auto a = x.data_ptr<float>();
    auto b = w.data_ptr<float>();
    auto c = dy.data_ptr<float>();
    auto irp = inv_rms.data_ptr<float>();
    auto dxp = dx.data_ptr<float>();
    auto dwp = dw.data_ptr<float>();
    if (nv == 1)
      hipLaunchKernelGGL((rmsnorm_bwd_reg_kernel<float, 1>), dim3(nwg),
                         dim3(BLOCK), 0, stream, a, b, c, irp, dxp, dwp, n_rows, D);
    else if (nv == 2)
      hipLaunchKernelGGL((rmsnorm_bwd_reg_kernel<float, 2>), dim3(nwg),
                         dim3(BLOCK), 0, stream, a, b, c, irp, dxp, dwp, n_rows, D);
    else if (nv == 4)
      hipLaunchKernelGGL((rmsnorm_bwd_reg_kernel<float, 4>), dim3(nwg),
                         dim3(BLOCK), 0, stream, a, b, c, irp, dxp, dwp, n_rows, D);
    else
      hipLaunchKernelGGL(rmsnorm_bwd_kernel<float>, dim3(nwg), dim3(BLOCK),
                         shmem, stream, a, b, c, irp, dxp, dwp, n_rows, D);
  } else {
    TORCH_CHECK(false, "rmsnorm_bwd: unsupported dtype");
  }
  return {dx, dw};
}
