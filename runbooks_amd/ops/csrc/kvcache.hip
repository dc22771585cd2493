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
    *reinterpret_cast<rb::bf16x8 *>(v_cache + dst) =
        *reinterpret_cast<const rb::bf16x8 *>(v + src);
  }
}

}  // namespace

void kv_append(at::Tensor k, at::Tensor v, at::Tensor k_cache, at::Tensor v_cache,
               at::Tensor slot_mapping) {
  TORCH_CHECK(k.is_cuda() && k.is_contiguous() && v.is_contiguous(), "kv_append: args");
  TORCH_CHECK(k.scalar_type() == at::kBFloat16 && k_cache.scalar_type() == at::kBFloat16,
              "kv_append: bf16 only");
  TORCH_CHECK(slot_mapping.scalar_type() == at::kInt, "kv_append: int32 slots");
  const int64_t n_tokens = k.size(0);
  const int hkv = (int)k_cache.size(1);
  const int bs = (int)k_cache.size(2);
  const int dh = (int)k_cache.size(3);
  TORCH_CHECK(dh % 8 == 0, "kv_append: Dh % 8 == 0");
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = rb::rb_grid_1d(n_tokens * hkv * (dh / 8), BLOCK);
  hipLaunchKernelGGL(kv_append_kernel, dim3(grid), dim3(BLOCK), 0, stream,
                     (const uint16_t *)k.data_ptr(), (const uint16_t *)v.data_ptr(),
                     (uint16_t *)k_cache.data_ptr(), (uint16_t *)v_cache.data_ptr(),
                     slot_mapping.data_ptr<int32_t>(), n_tokens, hkv, bs, dh);
}
