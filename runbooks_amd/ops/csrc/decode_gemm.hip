// Decode GEMM v2 for CDNA4 (gfx950): y[M,N] = x[M,K] @ W[N,K]^T,
// M <= 32 (decode batch), bf16 in / bf16 out, fp32 accumulate.
//
// The M<=32 decode GEMM is >99% W traffic, so the ONLY thing that
// matters is shaping the weight read as perfectly coalesced nontemporal
// bursts with enough in flight (MI355X ~6.3 TB/s achievable; hipBLASLt
// measured ~2.4 TB/s in the round-1 decode loop, and a fragment-shaped
// direct-load variant measured ~1.5 TB/s because every wave instruction
// became 32 scattered 32 B requests). This version therefore streams
// BOTH operands from layouts pre-swizzled into MFMA fragment-lane order:
//
//  * W is stored fragment-major at model-load time (decode_swizzle_w:
//    [N/32][K/16][lane][8] bf16) so each wave instruction is one 1 KiB
//    contiguous nontemporal read. 288 GB HBM3E per GPU makes the second
//    weight copy the right trade (same pattern as the fp8 registry);
//    prefill/training keep using the original [N,K] tensor via hipBLASLt.
//  * x (tiny: M*K <= 0.7 MB) is swizzled per call by decode_swizzle_x —
//    one extra ~microsecond kernel — into [K/16][lane][8] with zero
//    padding for m >= M; after that every wave reads the same contiguous
//    L2-resident stream.
//  * One wave owns 32 W rows x a DEEP contiguous k-range (K/(4*SPLIT)),
//    no LDS and no barrier in the main loop (hipcc pipelines the 2x8
//    ping-pong register ring freely; ~16 KB in flight per wave vs the
//    ~9 KB/CU Little's-law requirement at 24.6 GB/s/CU).
//  * 4 waves per WG take adjacent k-quarters; one 16 KB LDS reduction
//    combines them. SPLIT (1 for N>=16k, 2-4 for N=4096 shapes) adds
//    fp32 slabs [SPLIT,M,N] reduced by a combine kernel (~1 MB next to
//    the 32-90 MB W stream).
//
// Fragment maps (v_mfma_f32_32x32x16_bf16, A=W so C cols = m):
//   A lane l -> A[row=l&31][k=(l>>5)*8+i];  B lane l -> B[k][col=l&31];
//   C lane l -> C[row=(r&3)+8*(r>>2)+4*(l>>5)][col=l&31], r = 0..15.
// The swizzles bake exactly these maps into the storage order.
//
// Reference parity: replaces hipBLASLt for the decode hot loop the way
// the reference's external server image leans on cuBLAS
// (substratusai/runbooks docs/container-contract.md serving contract).

#include <stdlib.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;   // 4 waves
constexpr int MMAX = 32;

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

// One chunk of U k-steps. W nontemporal (streamed once, must not evict
// L2); x plain (re-read by every n-block: L2 is exactly where it wants
// to live). Both streams are lane*16B contiguous per instruction.
#define RB_LOAD_CHUNK(WB, XB, SBASE)                                   \
  _Pragma("unroll") for (int u = 0; u < U; ++u) {                      \
    WB[u] = nt_load8v(wseg + (int64_t)((SBASE) + u) * 512 + lane8);    \
    XB[u] = load8v(xs + (int64_t)((SBASE) + u) * 512 + lane8);         \
  }

#define RB_MFMA_CHUNK(WB, XB)                                          \
  _Pragma("unroll") for (int u = 0; u < U; ++u) {                      \
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(WB[u], XB[u], acc,   \
                                                  0, 0, 0);            \
  }

// STORE_BF16: write y bf16 directly (SPLIT == 1). Otherwise store an
// fp32 slab slice at slabs + kslice*M*N for decode_gemm_combine.
// U = k-steps (of 16) per unrolled chunk: 8 keeps 3 waves/SIMD, 12
// trades occupancy (2/SIMD) for a deeper in-flight ring (RB_DG_U=12).
template <bool STORE_BF16, int U = 8>
__global__ __launch_bounds__(BLOCK, 1) void decode_gemm_kernel(
    const uint16_t *__restrict__ xs, const uint16_t *__restrict__ ws,
    uint16_t *__restrict__ yp, float *__restrict__ slabs,
    int M, int N, int K) {
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;
  const int col = lane & 31;
  const int lane8 = lane * 8;

  const int nblk = blockIdx.x;           // 32 W rows per WG
  const int kslice = blockIdx.y;
  const int split = gridDim.y;

  // k-step range (steps of 16) for this (kslice, wave) group; contiguous
  // per group so the W stream is one long sequential burst.
  const int steps_total = K / 16;
  const int ngroups = split * 4;
  const int spg = (steps_total + ngroups - 1) / ngroups;
  const int g = kslice * 4 + wid;
  const int s0 = g * spg;
  const int s1 = min(steps_total, s0 + spg);

  const uint16_t *wseg = ws + (int64_t)nblk * steps_total * 512;

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
    bf16x8v w1 = nt_load8v(wseg + (int64_t)s * 512 + lane8);
    bf16x8v x1 = load8v(xs + (int64_t)s * 512 + lane8);
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

// slabs [SPLIT, M, N] f32 -> y [M, N] bf16
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

// x [M, K] row-major -> xs [K/16][64 lanes][8] bf16 (B-fragment lane
// order: lane = hi*32 + m, elems = x[m][s*16 + hi*8 + i]; zeros for
// m >= M). One scattered read pass of <= 0.7 MB, coalesced writes.
__global__ void decode_swizzle_x_kernel(
    const uint16_t *__restrict__ xp, uint16_t *__restrict__ xs,
    int M, int K) {
  const int steps = K / 16;
  const int64_t total = (int64_t)steps * 64;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += stride) {
    const int s = (int)(idx >> 6);
    const int l = (int)(idx & 63);
    const int m = l & 31;
    const int hi = l >> 5;
    rb::bf16x8 v;
    if (m < M) {
      v = *reinterpret_cast<const rb::bf16x8 *>(
          xp + (int64_t)m * K + s * 16 + hi * 8);
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) v.v[e] = 0;
    }
    *reinterpret_cast<rb::bf16x8 *>(xs + idx * 8) = v;
  }
}

}  // namespace

// Pick the cross-WG k-split by LOAD BALANCE: the kernel is a pure
// weight stream, so wall time = (WG waves over the 256 CUs) = ceil(
// nWG/256) rounds, and utilization = nWG / (256 * rounds). Measured
// r5 microbench tracks this exactly: 384 WGs -> 75% -> 4.2 TB/s,
// 688 -> 90% -> 4.7, 1000 -> 98% -> 5.2. Score each legal split and
// keep the best; ties go to fewer splits (a split adds a slab round
// trip + combine kernel ~5-6 us inside the captured graph).
int64_t decode_gemm_split(int64_t N, int64_t K) {
  // Split only when the grid cannot cover the 256 CUs (o_proj/down_proj
  // N=4096 -> 128 WGs). Measured r5/r6: splitting an already-covering
  // grid loses (gateup 688 WGs split 4: -10%; qkv 384 split 2 gained
  // ~2% on the kernel but pays a combine launch inside the graph).
  const int64_t nblocks = N / 32;
  int64_t split = 1;
  while (split < 8 && nblocks * split < 256 &&
         (K / 16) % (split * 4) == 0 && K / (split * 2) >= 1024) {
    split *= 2;
  }
  return split;
}

bool decode_gemm_supported(int64_t M, int64_t N, int64_t K) {
  return M >= 1 && M <= MMAX && N % 32 == 0 && K % 16 == 0 && K >= 1024;
}

// One-time weight swizzle (host-side tensor ops): [N,K] ->
// [N/32][K/16][2][32][8] = fragment-lane-major chunks of 1 KiB.
at::Tensor decode_swizzle_w(at::Tensor w) {
  TORCH_CHECK(w.dim() == 2 && w.scalar_type() == at::kBFloat16 &&
              w.is_contiguous(), "decode_swizzle_w: contiguous bf16 [N,K]");
  const int64_t N = w.size(0), K = w.size(1);
  TORCH_CHECK(N % 32 == 0 && K % 16 == 0, "decode_swizzle_w: shape");
  return w.view({N / 32, 32, K / 16, 2, 8})
      .permute({0, 2, 3, 1, 4}).contiguous();
}

at::Tensor decode_swizzle_x(at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous() &&
              x.scalar_type() == at::kBFloat16, "decode_swizzle_x: x");
  const int M = x.size(0), K = x.size(1);
  TORCH_CHECK(M <= MMAX && K % 16 == 0, "decode_swizzle_x: shape");
  auto xs = at::empty({(int64_t)(K / 16) * 512}, x.options());
  auto stream = at::cuda::getCurrentHIPStream();
  const int grid = rb::rb_grid_1d((int64_t)(K / 16) * 64, 256);
  hipLaunchKernelGGL(decode_swizzle_x_kernel, dim3(grid), dim3(256), 0,
                     stream, (const uint16_t *)x.data_ptr(),
                     (uint16_t *)xs.data_ptr(), M, K);
  return xs;
}

// Raw variant for fused consumers: split == 1 returns y bf16 [M, N];
// split > 1 returns the fp32 slab [split, M, N] UNCOMBINED — the
// consumer kernel (rmsnorm_res_slab_fwd_dec) folds the slices while it
// reads, saving the combine launch inside the decode graph.
static int dg_u() {
  static int u = [] {
    const char *e = getenv("RB_DG_U");
    return (e != nullptr && atoi(e) == 12) ? 12 : 8;
  }();
  return u;
}

at::Tensor decode_gemm_raw(at::Tensor xs, at::Tensor ws, int64_t M,
                           int64_t N, int64_t K) {
  TORCH_CHECK(xs.is_cuda() && ws.is_cuda() && xs.is_contiguous() &&
              ws.is_contiguous() && xs.scalar_type() == at::kBFloat16 &&
              ws.scalar_type() == at::kBFloat16, "decode_gemm_raw: inputs");
  TORCH_CHECK(xs.numel() == (K / 16) * 512 && ws.numel() == N * K &&
              decode_gemm_supported(M, N, K), "decode_gemm_raw: shape");
  auto stream = at::cuda::getCurrentHIPStream();
  const int split = (int)decode_gemm_split(N, K);
  if (split == 1) {
    auto y = at::empty({M, N}, xs.options());
    if (dg_u() == 12)
      hipLaunchKernelGGL((decode_gemm_kernel<true, 12>), dim3(N / 32, 1),
                         dim3(BLOCK), 0, stream,
                         (const uint16_t *)xs.data_ptr(),
                         (const uint16_t *)ws.data_ptr(),
                         (uint16_t *)y.data_ptr(), nullptr,
                         (int)M, (int)N, (int)K);
    else
      hipLaunchKernelGGL((decode_gemm_kernel<true, 8>), dim3(N / 32, 1),
                         dim3(BLOCK), 0, stream,
                         (const uint16_t *)xs.data_ptr(),
                         (const uint16_t *)ws.data_ptr(),
                         (uint16_t *)y.data_ptr(), nullptr,
                         (int)M, (int)N, (int)K);
    return y;
  }
  auto slabs = at::empty({split, M, N}, xs.options().dtype(at::kFloat));
  if (dg_u() == 12)
    hipLaunchKernelGGL((decode_gemm_kernel<false, 12>), dim3(N / 32, split),
                       dim3(BLOCK), 0, stream,
                       (const uint16_t *)xs.data_ptr(),
                       (const uint16_t *)ws.data_ptr(), nullptr,
                       (float *)slabs.data_ptr(), (int)M, (int)N, (int)K);
  else
    hipLaunchKernelGGL((decode_gemm_kernel<false, 8>), dim3(N / 32, split),
                       dim3(BLOCK), 0, stream,
                       (const uint16_t *)xs.data_ptr(),
                       (const uint16_t *)ws.data_ptr(), nullptr,
                       (float *)slabs.data_ptr(), (int)M, (int)N, (int)K);
  return slabs;
}

// xs from decode_swizzle_x, ws from decode_swizzle_w; M/N/K of the
// ORIGINAL y[M,N] = x[M,K] @ W[N,K]^T problem.
at::Tensor decode_gemm(at::Tensor xs, at::Tensor ws, int64_t M, int64_t N,
                       int64_t K, int64_t force_split) {
  TORCH_CHECK(xs.is_cuda() && ws.is_cuda() && xs.is_contiguous() &&
              ws.is_contiguous(), "decode_gemm: contiguous GPU tensors");
  TORCH_CHECK(xs.scalar_type() == at::kBFloat16 &&
              ws.scalar_type() == at::kBFloat16, "decode_gemm: bf16 only");
  TORCH_CHECK(xs.numel() == (K / 16) * 512, "decode_gemm: xs size");
  TORCH_CHECK(ws.numel() == N * K, "decode_gemm: ws size");
  TORCH_CHECK(decode_gemm_supported(M, N, K),
              "decode_gemm: unsupported shape ", M, "x", N, "x", K);

  auto stream = at::cuda::getCurrentHIPStream();
  auto y = at::empty({M, N}, xs.options());
  const int split = force_split > 0 ? (int)force_split
                                    : (int)decode_gemm_split(N, K);
  if (split == 1) {
    hipLaunchKernelGGL((decode_gemm_kernel<true>), dim3(N / 32, 1),
                       dim3(BLOCK), 0, stream,
                       (const uint16_t *)xs.data_ptr(),
                       (const uint16_t *)ws.data_ptr(),
                       (uint16_t *)y.data_ptr(), nullptr,
                       (int)M, (int)N, (int)K);
  } else {
    auto slabs = at::empty({split, M, N}, xs.options().dtype(at::kFloat));
    hipLaunchKernelGGL((decode_gemm_kernel<false>), dim3(N / 32, split),
                       dim3(BLOCK), 0, stream,
                       (const uint16_t *)xs.data_ptr(),
                       (const uint16_t *)ws.data_ptr(), nullptr,
                       (float *)slabs.data_ptr(), (int)M, (int)N, (int)K);
    const int64_t mn = M * N;
    const int grid = rb::rb_grid_1d(mn / 4, 256);
    hipLaunchKernelGGL(decode_gemm_combine_kernel, dim3(grid), dim3(256),
                       0, stream, (const float *)slabs.data_ptr(),
                       (uint16_t *)y.data_ptr(), split, mn);
  }
  return y;
}
