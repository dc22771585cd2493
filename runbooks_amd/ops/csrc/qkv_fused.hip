// Decode-path fusion epilogues for CDNA4 (gfx950).
//
// After the fused QKV GEMM (models/transformer.py fuse_for_inference) the
// eager path spent ~10% of serve kernel time on .contiguous() slice
// copies (profiles/). These kernels consume the packed GEMM outputs
// directly:
//
//  * qkv_rope_append: y [T, Hq*Dh + 2*Hkv*Dh] -> RoPE'd q (contiguous
//    out) + RoPE'd k and plain v appended straight into the paged KV
//    cache. Replaces 3 slice copies + 2 rope launches + kv_append.
//  * swiglu_packed: y [T, 2I] -> silu(y[:, :I]) * y[:, I:], replacing 2
//    slice copies + the 2-input swiglu kernel.
//
// RoPE math and cache layout match rope.hip / kvcache.hip (rotate-half,
// fp32 host tables; cache [blocks, Hkv, BS, Dh]).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;

__device__ __forceinline__ void rope_pair(
    const uint16_t *src1, const uint16_t *src2, uint16_t *dst1,
    uint16_t *dst2, const float *c, const float *s, int n) {
  for (int k = 0; k < n; ++k) {
    const float a = rb::bf16_to_f32(src1[k]);
    const float b = rb::bf16_to_f32(src2[k]);
    dst1[k] = rb::f32_to_bf16(a * c[k] - b * s[k]);
    dst2[k] = rb::f32_to_bf16(b * c[k] + a * s[k]);
  }
}

// VT: v_cache blocks transposed ([Dh, BS]) for the MFMA decode kernel
// (see kvcache.hip) — the v copy becomes 8 stride-BS element stores.
template <bool VT>
__global__ void qkv_rope_append_kernel(
    const uint16_t *__restrict__ y, uint16_t *__restrict__ q_out,
    uint16_t *__restrict__ k_cache, uint16_t *__restrict__ v_cache,
    const float *__restrict__ cs, const float *__restrict__ sn,
    const int32_t *__restrict__ pos, const int32_t *__restrict__ slots,
    int64_t T, int hq, int hkv, int dh, int bs) {
  constexpr int VP = 4;
  const int half = dh / 2;
  const int pv = half / VP;               // rope vec slots per head row
  const int vv = dh / 8;                  // copy slots per head row (v)
  const int ystride = (hq + 2 * hkv) * dh;
  const int64_t total_q = T * hq * pv;
  const int64_t total_k = T * hkv * pv;
  const int64_t total_v = T * hkv * vv;
  const int64_t total = total_q + total_k + total_v;
  const int64_t stride = (int64_t)gridDim.x * BLOCK;

  for (int64_t idx = (int64_t)blockIdx.x * BLOCK + threadIdx.x; idx < total;
       idx += stride) {
    if (idx < total_q) {                          // ---- q: rope -> q_out
      const int64_t row = idx / pv;               // t*hq + h
      const int v = (int)(idx % pv);
      const int64_t t = row / hq;
      const int h = (int)(row % hq);
      const int p = pos[t];
      const uint16_t *src = y + t * ystride + h * dh + v * VP;
      uint16_t *dst = q_out + row * dh + v * VP;
      rope_pair(src, src + half, dst, dst + half,
                cs + (int64_t)p * half + v * VP,
                sn + (int64_t)p * half + v * VP, VP);
    } else if (idx < total_q + total_k) {         // ---- k: rope -> cache
      const int64_t i = idx - total_q;
      const int64_t row = i / pv;
      const int v = (int)(i % pv);
      const int64_t t = row / hkv;
      const int h = (int)(row % hkv);
      const int slot = slots[t];
      if (slot < 0) continue;
      const int p = pos[t];
      const uint16_t *src = y + t * ystride + (hq + h) * dh + v * VP;
      uint16_t *dst = k_cache +
          ((((int64_t)(slot / bs) * hkv + h) * bs + slot % bs) * dh) + v * VP;
      rope_pair(src, src + half, dst, dst + half,
                cs + (int64_t)p * half + v * VP,
                sn + (int64_t)p * half + v * VP, VP);
    } else {                                      // ---- v: copy -> cache
      const int64_t i = idx - total_q - total_k;
      const int64_t row = i / vv;
      const int c = (int)(i % vv);
      const int64_t t = row / hkv;
      const int h = (int)(row % hkv);
      const int slot = slots[t];
      if (slot < 0) continue;
      const uint16_t *src = y + t * ystride + (hq + hkv + h) * dh + c * 8;
      if (VT) {
        const int64_t vbase =
            ((int64_t)(slot / bs) * hkv + h) * (int64_t)dh * bs;
        const rb::bf16x8 vv = *reinterpret_cast<const rb::bf16x8 *>(src);
#pragma unroll
        for (int e = 0; e < 8; ++e)
          v_cache[vbase + (int64_t)(c * 8 + e) * bs + slot % bs] = vv.v[e];
      } else {
        uint16_t *dst = v_cache +
            ((((int64_t)(slot / bs) * hkv + h) * bs + slot % bs) * dh) + c * 8;
        *reinterpret_cast<rb::bf16x8 *>(dst) =
            *reinterpret_cast<const rb::bf16x8 *>(src);
      }
    }
  }
}

template <typename T>
__global__ void swiglu_packed_kernel(const T *__restrict__ y,
                                     T *__restrict__ out,
                                     uint16_t *__restrict__ swz,
                                     int64_t n_rows, int inter) {
  constexpr int W = rb::VIO<T>::W;
  const int nvec = inter / W;
  const int64_t total = n_rows * nvec;
  const int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x; i < total;
       i += stride) {
    const int64_t row = i / nvec;
    const int v = (int)(i % nvec);
    float g[W], u[W];
    rb::VIO<T>::load(y + (row * 2 * inter) + v * W, g);
    rb::VIO<T>::load(y + (row * 2 * inter) + inter + v * W, u);
#pragma unroll
    for (int k = 0; k < W; ++k) {
      const float s = g[k] / (1.0f + __expf(-g[k]));
      g[k] = s * u[k];
    }
    rb::VIO<T>::store(out + row * inter + v * W, g);
    if constexpr (sizeof(T) == 2) {
      if (swz != nullptr) {
        const int k0 = v * 8;
        rb::VIO<T>::store(reinterpret_cast<T *>(swz) + (k0 >> 4) * 512 +
                          ((k0 >> 3) & 1) * 256 + row * 8, g);
      }
    }
  }
}

// gemma GeGLU: gelu_tanh(y[:, :I]) * y[:, I:]. Same memory shape as
// swiglu_packed; only the activation differs. Gated in python behind
// RB_FUSED_GEGLU until a GPU validation pass (round 2) - the eager
// torch path is the default for gelu_glu models.
template <typename T>
__global__ void geglu_packed_kernel(const T *__restrict__ y,
                                    T *__restrict__ out,
                                    uint16_t *__restrict__ swz,
                                    int64_t n_rows, int inter) {
  constexpr int W = rb::VIO<T>::W;
  const int nvec = inter / W;
  const int64_t total = n_rows * nvec;
  const int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x; i < total;
       i += stride) {
    const int64_t row = i / nvec;
    const int v = (int)(i % nvec);
    float g[W], u[W];
    rb::VIO<T>::load(y + (row * 2 * inter) + v * W, g);
    rb::VIO<T>::load(y + (row * 2 * inter) + inter + v * W, u);
#pragma unroll
    for (int k = 0; k < W; ++k) {
      // tanh approximation (HF gelu_pytorch_tanh)
      const float x = g[k];
      const float t = tanhf(0.7978845608028654f * (x + 0.044715f * x * x * x));
      g[k] = 0.5f * x * (1.0f + t) * u[k];
    }
    rb::VIO<T>::store(out + row * inter + v * W, g);
    if constexpr (sizeof(T) == 2) {
      if (swz != nullptr) {
        const int k0 = v * 8;
        rb::VIO<T>::store(reinterpret_cast<T *>(swz) + (k0 >> 4) * 512 +
                          ((k0 >> 3) & 1) * 256 + row * 8, g);
      }
    }
  }
}

}  // namespace

at::Tensor qkv_rope_append(at::Tensor y, at::Tensor cos, at::Tensor sin,
                           at::Tensor positions, at::Tensor k_cache,
                           at::Tensor v_cache, at::Tensor slot_mapping,
                           int64_t hq) {
  TORCH_CHECK(y.is_cuda() && y.is_contiguous() &&
              y.scalar_type() == at::kBFloat16, "qkv_rope_append: y");
  TORCH_CHECK(positions.scalar_type() == at::kInt &&
              slot_mapping.scalar_type() == at::kInt, "qkv_rope_append: idx");
  const int hkv = (int)k_cache.size(1);
  const int bs = (int)k_cache.size(2);
  const int dh = (int)k_cache.size(3);
  const int64_t T = y.size(0);
  TORCH_CHECK((int64_t)(hq + 2 * hkv) * dh == y.size(1),
              "qkv_rope_append: packed width mismatch");
  TORCH_CHECK((dh / 2) % 4 == 0, "qkv_rope_append: Dh/2 % 4");
  auto q = at::empty({T, hq, dh}, y.options());
  auto stream = at::hip::getCurrentHIPStream();
  const int64_t total = T * (hq + hkv) * (dh / 8) + T * hkv * (dh / 8);
  const int grid = rb::rb_grid_1d(total, BLOCK);
  const bool vt = v_cache.size(2) == (int64_t)dh &&
                  v_cache.size(3) == (int64_t)bs && dh != bs;
  if (vt) {
    hipLaunchKernelGGL(qkv_rope_append_kernel<true>, dim3(grid), dim3(BLOCK),
                       0, stream, (const uint16_t *)y.data_ptr(),
                       (uint16_t *)q.data_ptr(),
                       (uint16_t *)k_cache.data_ptr(),
                       (uint16_t *)v_cache.data_ptr(),
                       cos.data_ptr<float>(), sin.data_ptr<float>(),
                       positions.data_ptr<int32_t>(),
                       slot_mapping.data_ptr<int32_t>(),
                       T, (int)hq, hkv, dh, bs);
  } else {
    hipLaunchKernelGGL(qkv_rope_append_kernel<false>, dim3(grid), dim3(BLOCK),
                       0, stream, (const uint16_t *)y.data_ptr(),
                       (uint16_t *)q.data_ptr(),
                       (uint16_t *)k_cache.data_ptr(),
                       (uint16_t *)v_cache.data_ptr(),
                       cos.data_ptr<float>(), sin.data_ptr<float>(),
                       positions.data_ptr<int32_t>(),
                       slot_mapping.data_ptr<int32_t>(),
                       T, (int)hq, hkv, dh, bs);
  }
  return q;
}

at::Tensor swiglu_packed(at::Tensor y) {
  TORCH_CHECK(y.is_cuda() && y.is_contiguous(), "swiglu_packed: y");
  const int64_t inter2 = y.size(-1);
  TORCH_CHECK(inter2 % 2 == 0, "swiglu_packed: last dim even");
  const int inter = (int)(inter2 / 2);
  const int64_t n_rows = y.numel() / inter2;
  auto out = at::empty({n_rows, (int64_t)inter}, y.options());
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = rb::rb_grid_1d(n_rows * (inter / 8), BLOCK);
  if (y.scalar_type() == at::kBFloat16) {
    TORCH_CHECK(inter % 8 == 0, "swiglu_packed bf16: I % 8");
    hipLaunchKernelGGL(swiglu_packed_kernel<uint16_t>, dim3(grid),
                       dim3(BLOCK), 0, stream,
                       (const uint16_t *)y.data_ptr(),
                       (uint16_t *)out.data_ptr(), (uint16_t *)nullptr,
                       n_rows, inter);
  } else if (y.scalar_type() == at::kFloat) {
    TORCH_CHECK(inter % 4 == 0, "swiglu_packed f32: I % 4");
    hipLaunchKernelGGL(swiglu_packed_kernel<float>, dim3(grid), dim3(BLOCK),
                       0, stream, (const float *)y.data_ptr(),
                       (float *)out.data_ptr(), (uint16_t *)nullptr,
                       n_rows, inter);
  } else {
    TORCH_CHECK(false, "swiglu_packed: dtype");
  }
  return out;
}

at::Tensor geglu_packed(at::Tensor y) {
  TORCH_CHECK(y.is_cuda() && y.is_contiguous(), "geglu_packed: y");
  const int64_t inter2 = y.size(-1);
  TORCH_CHECK(inter2 % 2 == 0, "geglu_packed: last dim even");
  const int inter = (int)(inter2 / 2);
  const int64_t n_rows = y.numel() / inter2;
  auto out = at::empty({n_rows, (int64_t)inter}, y.options());
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = rb::rb_grid_1d(n_rows * (inter / 8), BLOCK);
  if (y.scalar_type() == at::kBFloat16) {
    TORCH_CHECK(inter % 8 == 0, "geglu_packed bf16: I % 8");
    hipLaunchKernelGGL(geglu_packed_kernel<uint16_t>, dim3(grid),
                       dim3(BLOCK), 0, stream,
                       (const uint16_t *)y.data_ptr(),
                       (uint16_t *)out.data_ptr(), (uint16_t *)nullptr,
                       n_rows, inter);
  } else if (y.scalar_type() == at::kFloat) {
    TORCH_CHECK(inter % 4 == 0, "geglu_packed f32: I % 4");
    hipLaunchKernelGGL(geglu_packed_kernel<float>, dim3(grid), dim3(BLOCK),
                       0, stream, (const float *)y.data_ptr(),
                       (float *)out.data_ptr(), (uint16_t *)nullptr,
                       n_rows, inter);
  } else {
    TORCH_CHECK(false, "geglu_packed: dtype");
  }
  return out;
}

// Decode variants: also emit the decode-GEMM pre-swizzled operand
// ([I/16]x512 bf16; rows m >= n_rows uninitialized — dropped by the
// GEMM epilogue). Saves the standalone x-swizzle launch per MLP.
static std::vector<at::Tensor> glu_packed_dec(at::Tensor y, bool gelu) {
  TORCH_CHECK(y.is_cuda() && y.is_contiguous() &&
              y.scalar_type() == at::kBFloat16, "glu_packed_dec: bf16");
  const int64_t inter2 = y.size(-1);
  const int inter = (int)(inter2 / 2);
  const int64_t n_rows = y.numel() / inter2;
  TORCH_CHECK(n_rows <= 32 && inter % 16 == 0, "glu_packed_dec: shape");
  auto sizes = y.sizes().vec();
  sizes.back() = inter;
  auto out = at::empty(sizes, y.options());
  auto swz = at::empty({(int64_t)(inter / 16) * 512}, y.options());
  auto stream = at::cuda::getCurrentHIPStream();
  const int grid = rb::rb_grid_1d(n_rows * (inter / 8), BLOCK);
  if (gelu) {
    hipLaunchKernelGGL(geglu_packed_kernel<uint16_t>, dim3(grid),
                       dim3(BLOCK), 0, stream,
                       (const uint16_t *)y.data_ptr(),
                       (uint16_t *)out.data_ptr(),
                       (uint16_t *)swz.data_ptr(), n_rows, inter);
  } else {
    hipLaunchKernelGGL(swiglu_packed_kernel<uint16_t>, dim3(grid),
                       dim3(BLOCK), 0, stream,
                       (const uint16_t *)y.data_ptr(),
                       (uint16_t *)out.data_ptr(),
                       (uint16_t *)swz.data_ptr(), n_rows, inter);
  }
  return {out, swz};
}

std::vector<at::Tensor> swiglu_packed_dec(at::Tensor y) {
  return glu_packed_dec(y, false);
}

std::vector<at::Tensor> geglu_packed_dec(at::Tensor y) {
  return glu_packed_dec(y, true);
}
