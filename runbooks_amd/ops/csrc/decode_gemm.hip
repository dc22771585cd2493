// Decode GEMM v2 for CDNA4 (gfx950): y[M,N] = x[M,K] @ W[N,K]^T,
// M <= 32 (decode batch), bf16 in / bf16 out, fp32 accumulate.
//
// Replaces the round-1 skinny_gemm tile design (64x256 LDS-staged tiles,
// K/256-way cross-WG split with fp32 slab round trips) with the shape the
// hardware actually wants for a pure weight stream (the M<=32 decode GEMM
// is >99% W traffic; MI355X_MICROARCH "GEMV / M <= 16 decode weights:
// load straight to VGPRs, deep unroll, late vmcnt"):
//
//  * One wave owns 32 W rows x a DEEP contiguous k-range (K/(4*SPLIT)),
//    so split-K slab traffic collapses: SPLIT is 1 for N>=8192 shapes and
//    2-4 for the N=4096 shapes (vs K/256 = 16..43 slices before).
//  * W streams straight into MFMA A-fragments with nontemporal 16 B/lane
//    loads (nt: streamed-once data must not displace L2/L1 — the
//    "nt-weights" row of the microarch price list). No LDS round trip,
//    no __syncthreads in the main loop, so hipcc pipelines the 2x8-deep
//    load ring freely (in-flight bytes per CU ~= waves x 16 x 1KB >> the
//    ~9 KB Little's-law requirement at 24.6 GB/s/CU).
//  * x fragments load from global per chunk: x is <=0.7 MB total and
//    L2/L3-resident, and the 32x32x16 MFMA shape halves x traffic per W
//    byte vs 16x16x32 (1 KB x per 1 KB W per instruction).
//  * 4 waves of a WG take adjacent k-quarters; one 16 KB LDS reduction
//    at the end combines them (still inside the workgroup: no
//    inter-workgroup visibility protocol needed). SPLIT>1 adds fp32
//    slabs [SPLIT,M,N] reduced by decode_gemm_combine (1-4 slabs, ~1 MB:
//    negligible next to the 32-90 MB W stream).
//
// Fragment maps (A = W rows so 16 B/lane stays row-contiguous; validated
// on-GPU by the round-1 skinny_gemm tests and test_decode_gemm):
//   v_mfma_f32_32x32x16_bf16: A lane l -> A[row=l&31][k=(l>>5)*8+i]
//   B lane l -> B[k=(l>>5)*8+i][col=l&31];  C lane l ->
//   C[row=(r&3)+8*(r>>2)+4*(l>>5)][col=l&31], r = 0..15.
// With A=W (row=n), B=x^T (col=m): acc[r] = y[m=l&31][n=n_local(r)].
//
// Reference parity: replaces hipBLASLt for the decode hot loop the way
// the reference's external server image relies on cuBLAS
// (substratusai/runbooks docs/container-contract.md serving contract).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;   // 4 waves
constexpr int MMAX = 32;
constexpr int U = 8;         // k-steps (of 16) per unrolled chunk

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(16))) float f32x16v;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4v;

__device__ __forceinline__ bf16x8v nt_load8v(const uint16_t *p) {
  union { u32x4v u; bf16x8v v; } c;
  c.u = __builtin_nontemporal_load(reinterpret_cast<const u32x4v *>(p));
  return c.v;
}

__device__ __forceinline__ bf16x8v load8v(const uint16_t *p) {
  union { u32x4v u; bf16x8v v; } c;
  c.u = *reinterpret_cast<const u32x4v *>(p);
  return c.v;
}

// One chunk of U k-steps: W via nt (A operand), x via plain load (B).
#define RB_LOAD_CHUNK(WB, XB, SBASE)                                   \
  _Pragma("unroll") for (int u = 0; u < U; ++u) {                      \
    WB[u] = nt_load8v(wrow + (int64_t)((SBASE) + u) * 16);             \
    XB[u] = load8v(xrow + (int64_t)((SBASE) + u) * 16);                \
  }

#define RB_MFMA_CHUNK(WB, XB)                                          \
  _Pragma("unroll") for (int u = 0; u < U; ++u) {                      \
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(WB[u], XB[u], acc,   \
                                                  0, 0, 0);            \
  }

// STORE_BF16: write y bf16 directly (SPLIT == 1). Otherwise store an
// fp32 slab slice at slab + kslice*M*N for decode_gemm_combine.
template <bool STORE_BF16>
__global__ __launch_bounds__(BLOCK, 1) void decode_gemm_kernel(
    const uint16_t *__restrict__ xp, const uint16_t *__restrict__ wp,
    uint16_t *__restrict__ yp, float *__restrict__ slabs,
    int M, int N, int K) {
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;
  const int col = lane & 31;

  const int nblk = blockIdx.x;           // 32 W rows per WG
  const int kslice = blockIdx.y;
  const int split = gridDim.y;

  // k-step range (steps of 16) for this (kslice, wave) group; contiguous
  // per group so each W row is one long sequential stream.
  const int steps_total = K / 16;
  const int ngroups = split * 4;
  const int spg = (steps_total + ngroups - 1) / ngroups;
  const int g = kslice * 4 + wid;
  const int s0 = g * spg;
  const int s1 = min(steps_total, s0 + spg);

  const int n = nblk * 32 + col;
  const int m = col;                     // B operand col = m
  // m >= M reads row M-1 (clamped, in-bounds); its outputs are dropped
  // in the epilogue, so no masking cost in the hot loop.
  const uint16_t *wrow = wp + (int64_t)n * K + hi * 8;
  const uint16_t *xrow = xp + (int64_t)min(m, M - 1) * K + hi * 8;

  f32x16v acc = (f32x16v)(0.0f);

  bf16x8v wA[U], xA[U], wB[U], xB[U];
  int s = s0;
  const int nmain = ((s1 - s0) / (2 * U)) * (2 * U);
  if (nmain > 0) {
    RB_LOAD_CHUNK(wA, xA, s);
    const int smain = s0 + nmain;
    for (; s + 2 * U <= smain; s += 2 * U) {
      if (s + U < smain) { RB_LOAD_CHUNK(wB, xB, s + U); }
      RB_MFMA_CHUNK(wA, xA);
      if (s + 2 * U < smain) { RB_LOAD_CHUNK(wA, xA, s + 2 * U); }
      if (s + U < smain) { RB_MFMA_CHUNK(wB, xB); }
    }
  }
  // tail: single k-steps, no prefetch (<= 2U-1 iterations)
  for (; s < s1; ++s) {
    bf16x8v w1 = nt_load8v(wrow + (int64_t)s * 16);
    bf16x8v x1 = load8v(xrow + (int64_t)s * 16);
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(w1, x1, acc, 0, 0, 0);
  }

  // ---- cross-wave reduction in LDS -------------------------------------
  __shared__ __attribute__((aligned(16))) float red[4][32][MMAX];
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int n_local = (r & 3) + 8 * (r >> 2) + 4 * hi;
    red[wid][n_local][col] = acc[r];
  }
  __syncthreads();

  // 1024 outputs, 4 per thread, n fastest so bf16 stores coalesce 8 B.
  const int em = tid >> 3;               // 0..31
  const int en = (tid & 7) * 4;          // 0,4,..,28
  if (em < M) {
    float sum[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      sum[j] = red[0][en + j][em] + red[1][en + j][em] +
               red[2][en + j][em] + red[3][en + j][em];
    }
    if (STORE_BF16) {
      union { uint16_t u[4]; uint64_t q; } o;
#pragma unroll
      for (int j = 0; j < 4; ++j) o.u[j] = rb::f32_to_bf16(sum[j]);
      *reinterpret_cast<uint64_t *>(
          yp + (int64_t)em * N + nblk * 32 + en) = o.q;
    } else {
      float *slab = slabs + ((int64_t)kslice * M + em) * N + nblk * 32 + en;
      *reinterpret_cast<float4 *>(slab) =
          make_float4(sum[0], sum[1], sum[2], sum[3]);
    }
  }
}

// slab [SPLIT, M, N] f32 -> y [M, N] bf16
__global__ void decode_gemm_combine_kernel(
    const float *__restrict__ slabs, uint16_t *__restrict__ yp,
    int split, int64_t mn) {
  const int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i = i0; i < mn; i += stride) {
    float4 s = *reinterpret_cast<const float4 *>(slabs + i);
    for (int k = 1; k < split; ++k) {
      float4 t = *reinterpret_cast<const float4 *>(slabs + (int64_t)k * mn + i);
      s.x += t.x; s.y += t.y; s.z += t.z; s.w += t.w;
    }
    union { uint16_t u[4]; uint64_t q; } o;
    o.u[0] = rb::f32_to_bf16(s.x); o.u[1] = rb::f32_to_bf16(s.y);
    o.u[2] = rb::f32_to_bf16(s.z); o.u[3] = rb::f32_to_bf16(s.w);
    *reinterpret_cast<uint64_t *>(yp + i) = o.q;
  }
}

}  // namespace

// Pick a cross-WG k-split so the grid covers the 256 CUs (>=2 WGs per CU
// where the shape allows; each WG is one CU-resident 4-wave block).
int64_t decode_gemm_split(int64_t N, int64_t K) {
  const int64_t nblocks = N / 32;
  int64_t split = 1;
  while (split < 8 && nblocks * split < 512 &&
         (K / 16) % (split * 2) == 0 && K / (split * 2) >= 512) {
    split *= 2;
  }
  return split;
}

bool decode_gemm_supported(int64_t M, int64_t N, int64_t K) {
  return M >= 1 && M <= MMAX && N % 32 == 0 && K % 16 == 0 && K >= 1024;
}

at::Tensor decode_gemm(at::Tensor x, at::Tensor w) {
  TORCH_CHECK(x.is_cuda() && w.is_cuda() && x.is_contiguous() &&
              w.is_contiguous(), "decode_gemm: contiguous GPU tensors");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16, "decode_gemm: bf16 only");
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK((int)w.size(1) == K, "decode_gemm: K mismatch");
  TORCH_CHECK(decode_gemm_supported(M, N, K),
              "decode_gemm: unsupported shape ", M, "x", N, "x", K);

  auto stream = at::cuda::getCurrentHIPStream();
  auto y = at::empty({M, N}, x.options());
  const int split = (int)decode_gemm_split(N, K);
  if (split == 1) {
    hipLaunchKernelGGL((decode_gemm_kernel<true>), dim3(N / 32, 1),
                       dim3(BLOCK), 0, stream,
                       (const uint16_t *)x.data_ptr(),
                       (const uint16_t *)w.data_ptr(),
                       (uint16_t *)y.data_ptr(), nullptr, M, N, K);
  } else {
    auto slabs = at::empty({split, M, N}, x.options().dtype(at::kFloat));
    hipLaunchKernelGGL((decode_gemm_kernel<false>), dim3(N / 32, split),
                       dim3(BLOCK), 0, stream,
                       (const uint16_t *)x.data_ptr(),
                       (const uint16_t *)w.data_ptr(), nullptr,
                       (float *)slabs.data_ptr(), M, N, K);
    const int64_t mn = (int64_t)M * N;
    const int grid = rb::rb_grid_1d(mn / 4, 256);
    hipLaunchKernelGGL(decode_gemm_combine_kernel, dim3(grid), dim3(256),
                       0, stream, (const float *)slabs.data_ptr(),
                       (uint16_t *)y.data_ptr(), split, mn);
  }
  return y;
}
