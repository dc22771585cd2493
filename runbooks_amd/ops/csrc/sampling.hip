// Token sampling kernels for CDNA4 (gfx950).
//
// greedy_sample: per-row argmax over the vocab (the serving engine's
// greedy decode — reference behavior: the external basaran server image's
// sampling loop, SURVEY.md §2b "server image").
//
// gumbel_sample: temperature sampling via the Gumbel-max trick:
// argmax(logits/T + G) with G ~ Gumbel(0,1) samples exactly from
// softmax(logits/T) — one memory-bound pass, no sort, no cumsum.
// Counter-based hash RNG keyed on (seed, row, col) → reproducible.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;

RB_DEV uint32_t hash_u32(uint32_t x) {
  x ^= x >> 16; x *= 0x7feb352dU;
  x ^= x >> 15; x *= 0x846ca68bU;
  x ^= x >> 16;
  return x;
}

// uniform in (0, 1]
RB_DEV float rng_uniform(uint64_t seed, uint32_t row, uint32_t col) {
  uint32_t h = hash_u32((uint32_t)seed ^ hash_u32(row * 0x9e3779b9U ^ hash_u32(col)));
  h ^= (uint32_t)(seed >> 32);
  h = hash_u32(h);
  return ((float)h + 1.0f) * (1.0f / 4294967296.0f);
}

template <typename T, bool GUMBEL>
__global__ void sample_kernel(const T *__restrict__ logits,
                              int32_t *__restrict__ out, int64_t n_rows, int V,
                              float inv_temp, uint64_t seed) {
  __shared__ float red_v[BLOCK / RB_WAVE];
  __shared__ int red_i[BLOCK / RB_WAVE];

  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const T *lr = logits + row * V;
    float best = -INFINITY;
    int best_i = 0;
    // vectorized main span (scalar bf16 loads are 2-2.5x slower and the
    // 125-deep per-thread scalar chain made this 36 us for [32, 32000])
    constexpr int W = rb::VIO<T>::W;
    const int nvec = V / W;
    for (int i = threadIdx.x; i < nvec; i += BLOCK) {
      float f[W];
      rb::VIO<T>::load(lr + i * W, f);
#pragma unroll
      for (int k = 0; k < W; ++k) {
        float v = f[k];
        const int idx = i * W + k;
        if (GUMBEL) {
          const float u = rng_uniform(seed, (uint32_t)row, (uint32_t)idx);
          v = v * inv_temp - __logf(-__logf(u));
        }
        if (v > best) { best = v; best_i = idx; }
      }
    }
    for (int i = nvec * W + threadIdx.x; i < V; i += BLOCK) {
      float v = rb::bf16_to_f32_or_id(lr[i]);
      if (GUMBEL) {
        const float u = rng_uniform(seed, (uint32_t)row, (uint32_t)i);
        v = v * inv_temp - __logf(-__logf(u));
      }
      if (v > best) { best = v; best_i = i; }
    }
    // wave argmax
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      const float ov = __shfl_xor(best, off, 64);
      const int oi = __shfl_xor(best_i, off, 64);
      if (ov > best || (ov == best && oi < best_i)) { best = ov; best_i = oi; }
    }
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    if (lane == 0) { red_v[wid] = best; red_i[wid] = best_i; }
    __syncthreads();
    if (threadIdx.x == 0) {
      float b = red_v[0]; int bi = red_i[0];
      for (int wv = 1; wv < BLOCK / RB_WAVE; ++wv)
        if (red_v[wv] > b || (red_v[wv] == b && red_i[wv] < bi)) { b = red_v[wv]; bi = red_i[wv]; }
      out[row] = bi;
    }
    __syncthreads();
  }
}

}  // namespace

at::Tensor sample_tokens(at::Tensor logits, double temperature, int64_t seed) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous(), "sample: contiguous GPU logits");
  const int V = (int)logits.size(-1);
  const int64_t n_rows = logits.numel() / V;
  auto out = at::empty({n_rows}, logits.options().dtype(at::kInt));
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = (int)std::min<int64_t>(n_rows, 2048);
  const bool greedy = temperature <= 0.0;
  const float inv_temp = greedy ? 1.0f : (float)(1.0 / temperature);

#define RB_SAMPLE_LAUNCH(T, G)                                                        \
  hipLaunchKernelGGL((sample_kernel<T, G>), dim3(grid), dim3(BLOCK), 0, stream,       \
                     (const T *)logits.data_ptr(), out.data_ptr<int32_t>(), n_rows,   \
                     V, inv_temp, (uint64_t)seed)

  if (logits.scalar_type() == at::kBFloat16) {
    if (greedy) RB_SAMPLE_LAUNCH(uint16_t, false); else RB_SAMPLE_LAUNCH(uint16_t, true);
  } else if (logits.scalar_type() == at::kFloat) {
    if (greedy) RB_SAMPLE_LAUNCH(float, false); else RB_SAMPLE_LAUNCH(float, true);
  } else {
    TORCH_CHECK(false, "sample: unsupported dtype");
  }
#undef RB_SAMPLE_LAUNCH
  return out;
}
