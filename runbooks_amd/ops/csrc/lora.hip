// Fused LoRA rank-r delta accumulate for CDNA4 (gfx950):
//
//   y[T, N] += scale * t[T, r] @ W[N, r]^T        (r <= 32)
//
// covers BOTH LoRA merge points of a train step (train/lora.py):
// forward  y += s * (x A^T) B^T   with t = x A^T, W = B
// backward dx += s * (dy B) A     with t = dy B,  W = A
//
// hipBLASLt runs these [2048, r<=32] @ [r, 4096..22016] accumulates at
// 12-20 us each (~700 calls per llama2-7b LoRA step = ~10% of the step,
// profiles/train_*_top_kernels.csv): tiny-K GEMMs are latency-bound in
// a library tiled for big K. Here the whole thing is one
// bandwidth-shaped pass: y is read+written once (the unavoidable
// traffic), t rows sit in registers, W panels are L2-resident (N*r*2 <=
// 1.4 MB), and the per-element work is r<=32 FMAs on the VALU — at
// [2048, 4096] r=16 the kernel is a pure 67 MB stream.
//
// Grid: one wave per 64-element y span; each lane owns one f32x?-wide
// column strip. No LDS, no barriers.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;
constexpr int RMAX = 32;

typedef __attribute__((ext_vector_type(4))) unsigned int u32x4v;

// WT: W stored [N, r] (y += t @ W^T, the B merge). !WT: W stored
// [r, N] (y += t @ W, the A merge in backward) — there each k reads a
// 16 B row chunk W[k, col8:col8+8], an even friendlier pattern.
template <bool WT, int R>
__global__ void lora_delta_kernel(const uint16_t *__restrict__ t,
                                  const uint16_t *__restrict__ w,
                                  uint16_t *__restrict__ y,
                                  int64_t T, int N, float scale) {
  constexpr int r = R;
  // element-vector id: 8 bf16 per thread
  const int64_t nvec_row = N / 8;
  const int64_t total = T * nvec_row;
  const int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (int64_t idx = (int64_t)blockIdx.x * BLOCK + threadIdx.x; idx < total;
       idx += stride) {
    const int64_t row = idx / nvec_row;
    const int col8 = (int)(idx % nvec_row) * 8;

    // t row (<= 32 bf16 = 64 B) — L2-hot, vectorized (scalar bf16
    // loads are the classic 2-2.5x CDNA4 trap and the first cut of this
    // kernel made it: 128 scalar loads/thread, train step 96 -> 208 ms)
    float tv[R];
    const uint16_t *tr = t + row * r;
#pragma unroll
    for (int k8 = 0; k8 < R / 8; ++k8) {
      float f[8];
      rb::VIO<uint16_t>::load(tr + k8 * 8, f);
#pragma unroll
      for (int e = 0; e < 8; ++e) tv[k8 * 8 + e] = f[e];
    }

    float acc[8];
    rb::bf16x8 yv = *reinterpret_cast<const rb::bf16x8 *>(y + row * N + col8);
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = rb::bf16_to_f32(yv.v[e]);

    if (WT) {
      // W rows col8..col8+7, r bf16 each (L2-resident panel), 16 B loads
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const uint16_t *wr = w + (int64_t)(col8 + e) * r;
        float s = 0.0f;
#pragma unroll
        for (int k8 = 0; k8 < R / 8; ++k8) {
          float f[8];
          rb::VIO<uint16_t>::load(wr + k8 * 8, f);
#pragma unroll
          for (int q = 0; q < 8; ++q) s += tv[k8 * 8 + q] * f[q];
        }
        acc[e] += scale * s;
      }
    } else {
      float s8[8] = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
      for (int k = 0; k < R; ++k) {
        rb::bf16x8 wv =
            *reinterpret_cast<const rb::bf16x8 *>(w + (int64_t)k * N + col8);
#pragma unroll
        for (int e = 0; e < 8; ++e)
          s8[e] += tv[k] * rb::bf16_to_f32(wv.v[e]);
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] += scale * s8[e];
    }
#pragma unroll
    for (int e = 0; e < 8; ++e) yv.v[e] = rb::f32_to_bf16(acc[e]);
    *reinterpret_cast<rb::bf16x8 *>(y + row * N + col8) = yv;
  }
}

// ---------------------------------------------------------------------------
// MFMA variant of the B merge: y[T, N] += scale * t[T, 16] @ W[N, 16]^T.
//
// hipBLASLt runs these K=16 accumulates at ~1.3 TB/s of y traffic (4x
// off the stream roofline — measured 68 us for [2048, 11008], r33
// torch.profiler). Unlike the scalar lora_delta_kernel above (per-lane
// W gather, measured slower in-step), BOTH operands here are natural
// 32x32x16 fragments read 16 B/lane CONTIGUOUSLY:
//   A[i=tok][k=r]  = t rows   (lane l: t[row0+l&31][(l>>5)*8+e])
//   B[k=r][j=n]    = W rows   (lane l: w[col0+l&31][(l>>5)*8+e])
// One MFMA per 32x32 y tile; C goes through a small per-wave LDS image
// so the y read-modify-write is row-coalesced. y traffic is the whole
// cost.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(8))) __bf16 rb_bf16x8w;
typedef __attribute__((ext_vector_type(16))) float rb_f32x16w;

__global__ __launch_bounds__(BLOCK) void lora_badd_kernel(
    const uint16_t *__restrict__ t, const uint16_t *__restrict__ w,
    uint16_t *__restrict__ y, int64_t T, int N, int nwtiles, float scale) {
  constexpr int R = 16;
  const int lane = (int)threadIdx.x & 63;
  const int wid = (int)threadIdx.x >> 6;
  const int c32 = lane & 31;
  const int h32 = lane >> 5;
  const int wt = blockIdx.x * 4 + wid;        // this wave's 32-col tile
  if (wt >= nwtiles) return;
  const int col0 = wt * 32;
  const int64_t row0 = (int64_t)blockIdx.y * 32;

  const int64_t trow = row0 + c32 < T ? row0 + c32 : T - 1;
  const rb_bf16x8w af = *reinterpret_cast<const rb_bf16x8w *>(
      t + trow * R + h32 * 8);
  const rb_bf16x8w bf = *reinterpret_cast<const rb_bf16x8w *>(
      w + (int64_t)(col0 + c32) * R + h32 * 8);
  rb_f32x16w c = (rb_f32x16w)(0.0f);
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, c, 0, 0, 0);

  __shared__ float lds[4][32][33];
#pragma unroll
  for (int r = 0; r < 16; ++r)
    lds[wid][(r & 3) + 8 * (r >> 2) + 4 * h32][c32] = c[r];
  // wave-private region: lane-lockstep ds ordering, no barrier needed

  if (row0 + c32 < T) {
    uint16_t *yr = y + (row0 + c32) * (int64_t)N + col0 + h32 * 16;
#pragma unroll
    for (int c8 = 0; c8 < 2; ++c8) {
      float f[8];
      rb::VIO<uint16_t>::load(yr + c8 * 8, f);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        f[e] += scale * lds[wid][c32][h32 * 16 + c8 * 8 + e];
      rb::VIO<uint16_t>::store(yr + c8 * 8, f);
    }
  }
}

}  // namespace

// MFMA B-merge: y += scale * t[T,16] @ w[N,16]^T (in place).
at::Tensor lora_badd_(at::Tensor y, at::Tensor t, at::Tensor w,
                      double scale) {
  TORCH_CHECK(y.is_cuda() && y.is_contiguous() && t.is_contiguous() &&
              w.is_contiguous(), "lora_badd_: contiguous GPU tensors");
  TORCH_CHECK(y.scalar_type() == at::kBFloat16 &&
              t.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16, "lora_badd_: bf16");
  const int64_t T = t.size(0);
  const int N = (int)w.size(0);
  TORCH_CHECK(t.size(1) == 16 && w.size(1) == 16 && y.size(0) == T &&
              (int)y.size(1) == N && N % 32 == 0, "lora_badd_: shape");
  auto stream = at::cuda::getCurrentHIPStream();
  const int nwtiles = N / 32;
  const dim3 grid((nwtiles + 3) / 4, (unsigned)((T + 31) / 32));
  hipLaunchKernelGGL(lora_badd_kernel, grid, dim3(BLOCK), 0, stream,
                     (const uint16_t *)t.data_ptr(),
                     (const uint16_t *)w.data_ptr(),
                     (uint16_t *)y.data_ptr(), T, N, nwtiles,
                     (float)scale);
  return y;
}

// In-place: y += scale * t @ w^T (w [N,r], w_transposed=true) or
// y += scale * t @ w (w [r,N], w_transposed=false). bf16, r <= 32.
at::Tensor lora_delta_(at::Tensor y, at::Tensor t, at::Tensor w,
                       double scale, bool w_transposed) {
  TORCH_CHECK(y.is_cuda() && y.is_contiguous() && t.is_contiguous() &&
              w.is_contiguous(), "lora_delta_: contiguous GPU tensors");
  TORCH_CHECK(y.scalar_type() == at::kBFloat16 &&
              t.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16, "lora_delta_: bf16");
  const int64_t T = t.size(0);
  const int r = (int)t.size(1);
  const int N = (int)(w_transposed ? w.size(0) : w.size(1));
  const int wr = (int)(w_transposed ? w.size(1) : w.size(0));
  TORCH_CHECK(r <= RMAX && r % 8 == 0 && wr == r && y.size(0) == T &&
              (int)y.size(1) == N && N % 8 == 0, "lora_delta_: shape");
  auto stream = at::cuda::getCurrentHIPStream();
  const int grid = rb::rb_grid_1d(T * (N / 8), BLOCK);
  auto tp = (const uint16_t *)t.data_ptr();
  auto wp = (const uint16_t *)w.data_ptr();
  auto yp = (uint16_t *)y.data_ptr();
  const float sc = (float)scale;
#define RB_LORA_LAUNCH(WTV, RV)                                         \
  hipLaunchKernelGGL((lora_delta_kernel<WTV, RV>), dim3(grid),          \
                     dim3(BLOCK), 0, stream, tp, wp, yp, T, N, sc)
  if (w_transposed) {
    if (r == 8) RB_LORA_LAUNCH(true, 8);
    else if (r == 16) RB_LORA_LAUNCH(true, 16);
    else if (r == 24) RB_LORA_LAUNCH(true, 24);
    else RB_LORA_LAUNCH(true, 32);
  } else {
    if (r == 8) RB_LORA_LAUNCH(false, 8);
    else if (r == 16) RB_LORA_LAUNCH(false, 16);
    else if (r == 24) RB_LORA_LAUNCH(false, 24);
    else RB_LORA_LAUNCH(false, 32);
  }
#undef RB_LORA_LAUNCH
  return y;
}
