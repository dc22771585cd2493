// Paged KV-cache append for CDNA4 (gfx950).
//
// Cache layout (chosen for decode-kernel coalescing and the 288 GB HBM3E
// budget — block granularity keeps fragmentation low while a [BS, Dh]
// inner tile keeps 4-token stripes contiguous for 16 B/lane reads):
//   k_cache, v_cache: [num_blocks, Hkv, BS, Dh]  (bf16)
//   slot_mapping:     [n_tokens] int32, slot = block_id * BS + offset
//
// One grid-stride pass; each (token, head) row is copied 16 B/lane.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;

// VT: v_cache blocks are TRANSPOSED ([Dh, BS] instead of [BS, Dh]) for
// the MFMA decode kernel's direct V^T fragment reads — the v write
// becomes 8 element stores at stride BS (append is write-once traffic,
// ~1000x smaller than the decode-side reads it speeds up).
template <bool VT>
__global__ void kv_append_kernel(const uint16_t *__restrict__ k,
                                 const uint16_t *__restrict__ v,
                                 uint16_t *__restrict__ k_cache,
                                 uint16_t *__restrict__ v_cache,
                                 const int32_t *__restrict__ slots,
                                 int64_t n_tokens, int hkv, int bs, int dh) {
  const int vec = dh / 8;                     // bf16x8 chunks per row
  const int64_t total = n_tokens * hkv * vec;
  const int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x; i < total; i += stride) {
    const int64_t tok = i / (hkv * vec);
    const int rem = (int)(i % (hkv * vec));
    const int h = rem / vec;
    const int c = rem % vec;
    const int slot = slots[tok];
    if (slot < 0) continue;                   // padding token
    const int blk = slot / bs;
    const int off = slot % bs;
    const int64_t src = (tok * hkv + h) * (int64_t)dh + c * 8;
    const int64_t dst = (((int64_t)blk * hkv + h) * bs + off) * dh + c * 8;
    *reinterpret_cast<rb::bf16x8 *>(k_cache + dst) =
        *reinterpret_cast<const rb::bf16x8 *>(k + src);
    if (VT) {
      const int64_t vbase = ((int64_t)blk * hkv + h) * (int64_t)dh * bs;
      const rb::bf16x8 vv = *reinterpret_cast<const rb::bf16x8 *>(v + src);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        v_cache[vbase + (int64_t)(c * 8 + e) * bs + off] = vv.v[e];
    } else {
      *reinterpret_cast<rb::bf16x8 *>(v_cache + dst) =
          *reinterpret_cast<const rb::bf16x8 *>(v + src);
    }
  }
}

// fp8-e4m3 append: one 16-lane group quantizes one (token, head) row
// for k and v each — absmax over Dh via group shfl-reduce, scale =
// absmax/448 stored after the Dh bytes (row stride Dh + 8).
// VT: the v cache is transposed e4m3 ([dh+4][bs] bytes per block; the
// 4 tail rows are the 16 per-token f32 scales) for the fp8 MFMA decode
// path. k rows keep the scalar layout either way.
template <bool VT>
__global__ void kv_append_fp8_kernel(const uint16_t *__restrict__ k,
                                     const uint16_t *__restrict__ v,
                                     uint8_t *__restrict__ k_cache,
                                     uint8_t *__restrict__ v_cache,
                                     const int32_t *__restrict__ slots,
                                     int64_t n_tokens, int hkv, int bs,
                                     int dh) {
  const int ve = dh / 16;                   // elems per lane (<= 16)
  const int rb8 = dh + 16;
  const int64_t units = n_tokens * hkv * 2; // (token, head, k-or-v)
  const int64_t groups_per_block = BLOCK / 16;
  const int gl = threadIdx.x & 15;
  const int64_t g0 = (int64_t)blockIdx.x * groups_per_block +
                     (threadIdx.x >> 4);
  const int64_t gstride = (int64_t)gridDim.x * groups_per_block;
  for (int64_t u = g0; u < units; u += gstride) {
    const int is_v = (int)(u & 1);
    const int64_t th = u >> 1;
    const int64_t tok = th / hkv;
    const int h = (int)(th % hkv);
    const int slot = slots[tok];
    if (slot < 0) continue;
    const uint16_t *src = (is_v ? v : k) + (tok * hkv + h) * (int64_t)dh +
                          gl * ve;
    float f[16];
    float amax = 1e-8f;
    for (int e = 0; e < ve; ++e) {
      f[e] = rb::bf16_to_f32(src[e]);
      amax = fmaxf(amax, fabsf(f[e]));
    }
#pragma unroll
    for (int off = 8; off > 0; off >>= 1)
      amax = fmaxf(amax, __shfl_xor(amax, off, 64));
    const float scale = amax / 448.0f;
    const float inv = 1.0f / scale;
    if (VT && is_v) {
      const int off = slot % bs;
      uint8_t *vb = v_cache +
          ((int64_t)(slot / bs) * hkv + h) * (int64_t)(dh + 4) * bs;
      for (int e = 0; e < ve; ++e)
        vb[(int64_t)(gl * ve + e) * bs + off] = rb::f32_to_fp8(f[e] * inv);
      if (gl == 0)
        *reinterpret_cast<float *>(vb + (int64_t)dh * bs + off * 4) = scale;
    } else {
      uint8_t *row = (is_v ? v_cache : k_cache) +
          (((int64_t)(slot / bs) * hkv + h) * bs + slot % bs) * rb8;
      for (int e = 0; e < ve; ++e)
        row[gl * ve + e] = rb::f32_to_fp8(f[e] * inv);
      if (gl == 0) *reinterpret_cast<float *>(row + dh) = scale;
    }
  }
}

}  // namespace

void kv_append(at::Tensor k, at::Tensor v, at::Tensor k_cache, at::Tensor v_cache,
               at::Tensor slot_mapping) {
  TORCH_CHECK(k.is_cuda() && k.is_contiguous() && v.is_contiguous(), "kv_append: args");
  TORCH_CHECK(k.scalar_type() == at::kBFloat16, "kv_append: k bf16");
  TORCH_CHECK(slot_mapping.scalar_type() == at::kInt, "kv_append: int32 slots");
  const int64_t n_tokens = k.size(0);
  const int hkv = (int)k_cache.size(1);
  const int bs = (int)k_cache.size(2);
  auto stream = at::hip::getCurrentHIPStream();
  if (k_cache.scalar_type() == at::kByte) {
    const int dh = (int)k_cache.size(3) - 16;
    TORCH_CHECK(dh % 16 == 0 && dh <= 256, "kv_append fp8: Dh % 16");
    const bool vt = v_cache.size(2) == (int64_t)(dh + 4) &&
                    v_cache.size(3) == (int64_t)bs;
    const int grid = rb::rb_grid_1d(n_tokens * hkv * 2 * 16, BLOCK);
    if (vt) {
      hipLaunchKernelGGL(kv_append_fp8_kernel<true>, dim3(grid), dim3(BLOCK),
                         0, stream, (const uint16_t *)k.data_ptr(),
                         (const uint16_t *)v.data_ptr(),
                         (uint8_t *)k_cache.data_ptr(),
                         (uint8_t *)v_cache.data_ptr(),
                         slot_mapping.data_ptr<int32_t>(), n_tokens, hkv, bs,
                         dh);
    } else {
      hipLaunchKernelGGL(kv_append_fp8_kernel<false>, dim3(grid), dim3(BLOCK),
                         0, stream, (const uint16_t *)k.data_ptr(),
                         (const uint16_t *)v.data_ptr(),
                         (uint8_t *)k_cache.data_ptr(),
                         (uint8_t *)v_cache.data_ptr(),
                         slot_mapping.data_ptr<int32_t>(), n_tokens, hkv, bs,
                         dh);
    }
    return;
  }
  TORCH_CHECK(k_cache.scalar_type() == at::kBFloat16, "kv_append: bf16 cache");
  const int dh = (int)k_cache.size(3);
  TORCH_CHECK(dh % 8 == 0, "kv_append: Dh % 8 == 0");
  const bool vt = v_cache.size(2) == (int64_t)dh &&
                  v_cache.size(3) == (int64_t)bs && dh != bs;
  const int grid = rb::rb_grid_1d(n_tokens * hkv * (dh / 8), BLOCK);
  if (vt) {
    hipLaunchKernelGGL(kv_append_kernel<true>, dim3(grid), dim3(BLOCK), 0,
                       stream, (const uint16_t *)k.data_ptr(),
                       (const uint16_t *)v.data_ptr(),
                       (uint16_t *)k_cache.data_ptr(),
                       (uint16_t *)v_cache.data_ptr(),
                       slot_mapping.data_ptr<int32_t>(), n_tokens, hkv, bs, dh);
  } else {
    hipLaunchKernelGGL(kv_append_kernel<false>, dim3(grid), dim3(BLOCK), 0,
                       stream, (const uint16_t *)k.data_ptr(),
                       (const uint16_t *)v.data_ptr(),
                       (uint16_t *)k_cache.data_ptr(),
                       (uint16_t *)v_cache.data_ptr(),
                       slot_mapping.data_ptr<int32_t>(), n_tokens, hkv, bs, dh);
  }
}
