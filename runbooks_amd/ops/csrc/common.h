// Common device helpers for runbooks_amd CDNA4 (gfx950) kernels.
//
// Written MI355X-first: wave64 everywhere, vectorized bf16 access in
// 8/16-byte units (hipcc does not auto-vectorize bf16 loads), grid-stride
// loops capped so the scheduler has room (256 CUs x 8 XCDs).
#pragma once

#include <hip/hip_fp8.h>
#include <hip/hip_runtime.h>
#include <stdint.h>

#define RB_WAVE 64
#define RB_DEV __device__ __forceinline__

namespace rb {

// ---------------------------------------------------------------------------
// bf16 <-> f32 (explicit bit ops; avoids header/format portability issues)
// ---------------------------------------------------------------------------
RB_DEV float bf16_to_f32(uint16_t u) {
  union { uint32_t u32; float f; } c;
  c.u32 = (uint32_t)u << 16;
  return c.f;
}

RB_DEV uint16_t f32_to_bf16(float f) {
  union { float f; uint32_t u32; } c;
  c.f = f;
  uint32_t u = c.u32;
  // round-to-nearest-even
  uint32_t rounding_bias = 0x7FFF + ((u >> 16) & 1);
  u += rounding_bias;
  return (uint16_t)(u >> 16);
}

// 8 x bf16 = 16 bytes: the coalescing sweet spot for wave64 streams.
struct bf16x8 { uint16_t v[8]; };

// ---------------------------------------------------------------------------
// OCP fp8-e4m3 (KV-cache compression; gfx950 has native cvt both ways)
// ---------------------------------------------------------------------------
RB_DEV uint8_t f32_to_fp8(float f) {
  __hip_fp8_e4m3 v(f);
  return v.__x;
}

// 8 packed e4m3 (one u64) -> 8 f32 via v_cvt_pk_f32_fp8
RB_DEV void fp8x8_to_f32(const uint8_t *p, float *f) {
  const uint32_t w0 = *reinterpret_cast<const uint32_t *>(p);
  const uint32_t w1 = *reinterpret_cast<const uint32_t *>(p + 4);
  auto a = __builtin_amdgcn_cvt_pk_f32_fp8(w0, false);
  auto b = __builtin_amdgcn_cvt_pk_f32_fp8(w0, true);
  auto c = __builtin_amdgcn_cvt_pk_f32_fp8(w1, false);
  auto d = __builtin_amdgcn_cvt_pk_f32_fp8(w1, true);
  f[0] = a[0]; f[1] = a[1]; f[2] = b[0]; f[3] = b[1];
  f[4] = c[0]; f[5] = c[1]; f[6] = d[0]; f[7] = d[1];
}

// 4 packed e4m3 (one u32) -> 4 f32
RB_DEV void fp8x4_to_f32(const uint8_t *p, float *f) {
  const uint32_t w = *reinterpret_cast<const uint32_t *>(p);
  auto a = __builtin_amdgcn_cvt_pk_f32_fp8(w, false);
  auto b = __builtin_amdgcn_cvt_pk_f32_fp8(w, true);
  f[0] = a[0]; f[1] = a[1]; f[2] = b[0]; f[3] = b[1];
}

// one u32 of packed e4m3 (already in a register) -> 4 f32; lets a
// pipelined consumer keep raw bytes in registers and convert at use
RB_DEV void fp8w_to_f32(uint32_t w, float *f) {
  auto a = __builtin_amdgcn_cvt_pk_f32_fp8(w, false);
  auto b = __builtin_amdgcn_cvt_pk_f32_fp8(w, true);
  f[0] = a[0]; f[1] = a[1]; f[2] = b[0]; f[3] = b[1];
}

// raw vector load of RW u32 words (8/16/32 B) — no conversion, so the
// compiler can keep the s_waitcnt away from the issue point
template <int RW>
RB_DEV void ld_words(uint32_t *dst, const uint8_t *src) {
  typedef __attribute__((ext_vector_type(2))) unsigned int u32x2i;
  typedef __attribute__((ext_vector_type(4))) unsigned int u32x4i;
  if constexpr (RW == 1) {
    *dst = *reinterpret_cast<const uint32_t *>(src);
  } else if constexpr (RW == 2) {
    *reinterpret_cast<u32x2i *>(dst) = *reinterpret_cast<const u32x2i *>(src);
  } else if constexpr (RW == 4) {
    *reinterpret_cast<u32x4i *>(dst) = *reinterpret_cast<const u32x4i *>(src);
  } else {
    static_assert(RW == 8, "unsupported row width");
    reinterpret_cast<u32x4i *>(dst)[0] =
        reinterpret_cast<const u32x4i *>(src)[0];
    reinterpret_cast<u32x4i *>(dst)[1] =
        reinterpret_cast<const u32x4i *>(src)[1];
  }
}

// 16 packed e4m3 (one 16 B vector) -> 16 f32
RB_DEV void fp8x16_to_f32(const uint8_t *p, float *f) {
  typedef __attribute__((ext_vector_type(4))) unsigned int u32x4i;
  const u32x4i w = *reinterpret_cast<const u32x4i *>(p);
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    auto a = __builtin_amdgcn_cvt_pk_f32_fp8(w[i], false);
    auto b = __builtin_amdgcn_cvt_pk_f32_fp8(w[i], true);
    f[i * 4 + 0] = a[0]; f[i * 4 + 1] = a[1];
    f[i * 4 + 2] = b[0]; f[i * 4 + 3] = b[1];
  }
}

// fp8 KV-cache row layout: dh e4m3 bytes, the f32 scale, pad to 16 B
// (16-aligned rows keep the decode loop on full 16 B/lane loads —
// 8 B accesses run at 0.54-0.70x the 16 B rate on gfx950).
RB_DEV constexpr int fp8_row_bytes(int dh) { return dh + 16; }
struct f32x8 { float v[8]; };

RB_DEV f32x8 to_f32(const bf16x8 &a) {
  f32x8 r;
#pragma unroll
  for (int i = 0; i < 8; ++i) r.v[i] = bf16_to_f32(a.v[i]);
  return r;
}

RB_DEV bf16x8 to_bf16(const f32x8 &a) {
  bf16x8 r;
#pragma unroll
  for (int i = 0; i < 8; ++i) r.v[i] = f32_to_bf16(a.v[i]);
  return r;
}

// Scalar dtype-generic helpers (bf16-as-u16 or f32).
RB_DEV float bf16_to_f32_or_id(uint16_t u) { return bf16_to_f32(u); }
RB_DEV float bf16_to_f32_or_id(float f) { return f; }
RB_DEV void store_scalar(uint16_t *p, float f) { *p = f32_to_bf16(f); }
RB_DEV void store_scalar(float *p, float f) { *p = f; }

// ---------------------------------------------------------------------------
// Wave-level reductions (64 lanes).
// ---------------------------------------------------------------------------
RB_DEV float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, 64);
  return x;
}

RB_DEV float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, 64));
  return x;
}

// Block reduction via LDS; `tmp` must hold >= blockDim.x/64 floats.
// Result is valid on every thread.
RB_DEV float block_reduce_sum(float x, float *tmp) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nwaves = (blockDim.x + 63) >> 6;
  x = wave_reduce_sum(x);
  if (lane == 0) tmp[wid] = x;
  __syncthreads();
  float r = (lane < nwaves) ? tmp[lane] : 0.0f;
  r = wave_reduce_sum(r);   // nwaves <= 16, fits one wave
  __syncthreads();
  return r;
}

RB_DEV float block_reduce_max(float x, float *tmp) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nwaves = (blockDim.x + 63) >> 6;
  x = wave_reduce_max(x);
  if (lane == 0) tmp[wid] = x;
  __syncthreads();
  float r = (lane < nwaves) ? tmp[lane] : -INFINITY;
  r = wave_reduce_max(r);
  __syncthreads();
  return r;
}

// Grid sizing for memory-bound grid-stride kernels: enough blocks to fill
// 256 CUs x several blocks each, capped so tail effects stay small.
static inline int rb_grid_1d(int64_t total, int block) {
  int64_t blocks = (total + block - 1) / block;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

}  // namespace rb

// ---------------------------------------------------------------------------
// Vectorized IO traits: bf16 in 16B (8 elems), f32 in 16B (4 elems).
// Both move 16 bytes per lane per instruction (global_load_dwordx4).
// ---------------------------------------------------------------------------
namespace rb {

template <typename T> struct VIO;

template <> struct VIO<uint16_t> {  // bf16 carried as raw u16
  static constexpr int W = 8;
  RB_DEV static void load(const uint16_t *p, float *f) {
    bf16x8 v = *reinterpret_cast<const bf16x8 *>(p);
#pragma unroll
    for (int i = 0; i < 8; ++i) f[i] = bf16_to_f32(v.v[i]);
  }
  RB_DEV static void store(uint16_t *p, const float *f) {
    bf16x8 v;
#pragma unroll
    for (int i = 0; i < 8; ++i) v.v[i] = f32_to_bf16(f[i]);
    *reinterpret_cast<bf16x8 *>(p) = v;
  }
};

template <> struct VIO<float> {
  static constexpr int W = 4;
  RB_DEV static void load(const float *p, float *f) {
    float4 v = *reinterpret_cast<const float4 *>(p);
    f[0] = v.x; f[1] = v.y; f[2] = v.z; f[3] = v.w;
  }
  RB_DEV static void store(float *p, const float *f) {
    *reinterpret_cast<float4 *>(p) = make_float4(f[0], f[1], f[2], f[3]);
  }
};

}  // namespace rb
