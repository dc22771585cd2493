// Skinny decode GEMM for CDNA4 (gfx950): y[M,N] = x[M,K] @ W[N,K]^T,
// M <= 32 (decode batch), bf16 in / bf16 out, fp32 accumulate.
//
// Why: batch-<=32 decode GEMMs are pure weight streams (W is ~99.9% of
// the traffic) and hipBLASLt's tiles measured only 25-45% of HBM
// bandwidth on these shapes (profiles/). Design:
//
//  * grid = (N/128 n-blocks) x KSPLIT k-slices -> >= 512 WGs on every
//    llama/falcon shape. WG = 4 waves; wave owns 32 W rows.
//  * x^T slice staged once to LDS in B-fragment-ready layout.
//  * main loop unrolled 8x: 8 independent nontemporal A-fragment loads
//    (16 B/lane, perfectly coalesced, nt keeps the one-pass W stream out
//    of L2 — MI355X_MICROARCH.md nt-weights) issued back-to-back, then 8
//    ds_read_b128 + 8 v_mfma_f32_32x32x16_bf16. ~8 loads in flight per
//    wave x 8+ waves/CU covers the ~900-cycle HBM latency
//    (cdna_hip_programming.md Guideline 7).
//  * split-K partials go to fp32 slabs with PLAIN stores; the kernel
//    boundary is the release, and a small combine kernel reduces
//    KSPLIT slabs -> bf16 y (no in-kernel cross-WG sync: the
//    boundary costs ~1.4 us, an agent-scope fence storm costs far more
//    — MI355X_MICROARCH.md boundary vs splitk-seam rows).
//
// mfma_f32_32x32x16_bf16 layouts as in attention_prefill.hip.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;        // 4 waves
constexpr int NB = 128;           // W rows per WG (32 per wave)
constexpr int MMAX = 32;
constexpr int UNROLL = 8;         // 16-k steps in flight

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(16))) float f32x16v;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4v;

__device__ __forceinline__ bf16x8v nt_load_frag(const uint16_t *p) {
  union { u32x4v u; bf16x8v v; } c;
  c.u = __builtin_nontemporal_load(reinterpret_cast<const u32x4v *>(p));
  return c.v;
}

// slabs layout: [nblk][kslice][NB][MMAX] f32
__global__ __launch_bounds__(BLOCK, 4) void skinny_gemm_kernel(
    const uint16_t *__restrict__ xp, const uint16_t *__restrict__ wp,
    float *__restrict__ slabs, int M, int N, int K, int ksplit) {
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;
  const int col = lane & 31;

  const int nblk = blockIdx.x;
  const int kslice = blockIdx.y;
  const int kper = K / ksplit;
  const int k0 = kslice * kper;

  // ---- stage x^T slice in B-fragment-ready layout -----------------------
  // slot s = (16-k step)*64 + lane: 8 contiguous k of x row (lane&31).
  extern __shared__ __attribute__((aligned(16))) uint16_t xfrag[];
  {
    const int slots = (kper / 16) * 64;
    for (int s = tid; s < slots; s += BLOCK) {
      const int ks = s >> 6;
      const int l = s & 63;
      const int m = l & 31;
      const int k = k0 + ks * 16 + (l >> 5) * 8;
      rb::bf16x8 v;
      if (m < M) {
        v = *reinterpret_cast<const rb::bf16x8 *>(xp + (int64_t)m * K + k);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) v.v[e] = 0;
      }
      *reinterpret_cast<rb::bf16x8 *>(xfrag + s * 8) = v;
    }
  }
  __syncthreads();

  // ---- main loop: UNROLL A loads in flight, then the MFMAs --------------
  const uint16_t *wrow =
      wp + (int64_t)(nblk * NB + wid * 32 + col) * K + k0 + hi * 8;
  f32x16v acc = (f32x16v)(0.0f);
  const int steps = kper / 16;
  int s = 0;
  for (; s + UNROLL <= steps; s += UNROLL) {
    bf16x8v a[UNROLL];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u)
      a[u] = nt_load_frag(wrow + (s + u) * 16);
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      const bf16x8v b = *reinterpret_cast<const bf16x8v *>(
          xfrag + (s + u) * 512 + (lane << 3));
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a[u], b, acc, 0, 0, 0);
    }
  }
  for (; s < steps; ++s) {
    const bf16x8v a = nt_load_frag(wrow + s * 16);
    const bf16x8v b = *reinterpret_cast<const bf16x8v *>(
        xfrag + s * 512 + (lane << 3));
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
  }

  // ---- write this slice's partial tile (plain stores; the kernel
  // boundary orders them before the combine kernel) -----------------------
  float *slab = slabs + (((int64_t)nblk * ksplit + kslice) * NB) * MMAX;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int n_local = wid * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
    slab[n_local * MMAX + col] = acc[r];
  }
}

// Combine KSPLIT fp32 slabs -> y bf16. Grid-strided over N*M.
__global__ void skinny_combine_kernel(const float *__restrict__ slabs,
                                      uint16_t *__restrict__ yp,
                                      int M, int N, int ksplit) {
  const int64_t total = (int64_t)N * M;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int n = (int)(i / M);
    const int m = (int)(i % M);
    const int nblk = n / NB;
    const int n_local = n % NB;
    const float *base =
        slabs + ((int64_t)nblk * ksplit * NB) * MMAX + n_local * MMAX + m;
    float v = 0.0f;
    for (int ks = 0; ks < ksplit; ++ks) v += base[(int64_t)ks * NB * MMAX];
    yp[(int64_t)m * N + n] = rb::f32_to_bf16(v);
  }
}

// Workspace cache (stable pointers across hipGraph replays).
struct SkinnyWorkspace {
  at::Tensor slabs;
};
SkinnyWorkspace &ws_for(const at::Tensor &ref, int64_t need) {
  static SkinnyWorkspace ws;
  if (!ws.slabs.defined() || ws.slabs.numel() < need ||
      ws.slabs.device() != ref.device()) {
    ws.slabs = at::empty({need}, ref.options().dtype(at::kFloat));
  }
  return ws;
}

}  // namespace

int64_t skinny_gemm_mmax() { return MMAX; }

at::Tensor skinny_gemm(at::Tensor x, at::Tensor w) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous(),
              "skinny_gemm: contiguous GPU tensors");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16, "skinny_gemm: bf16 only");
  const int M = (int)x.size(0), K = (int)x.size(1), N = (int)w.size(0);
  TORCH_CHECK(M <= MMAX, "skinny_gemm: M must be <= ", MMAX);
  TORCH_CHECK((int)w.size(1) == K, "skinny_gemm: K mismatch");
  TORCH_CHECK(N % NB == 0, "skinny_gemm: N % 128");
  TORCH_CHECK(K % 16 == 0, "skinny_gemm: K % 16");

  const int nblocks = N / NB;
  int ksplit = 1;
  while (nblocks * ksplit < 512 && ksplit < 16 &&
         (K / (ksplit * 2)) % 16 == 0 && K / (ksplit * 2) >= 128)
    ksplit *= 2;

  auto y = at::empty({M, N}, x.options());
  auto &ws = ws_for(x, (int64_t)nblocks * ksplit * NB * MMAX);
  auto stream = at::hip::getCurrentHIPStream();
  const size_t shmem = (size_t)(K / ksplit) * MMAX * sizeof(uint16_t);
  TORCH_CHECK(shmem <= 160 * 1024, "skinny_gemm: K/ksplit too large");
  hipLaunchKernelGGL(skinny_gemm_kernel, dim3(nblocks, ksplit), dim3(BLOCK),
                     shmem, stream,
                     (const uint16_t *)x.data_ptr(),
                     (const uint16_t *)w.data_ptr(),
                     ws.slabs.data_ptr<float>(), M, N, K, ksplit);
  const int cgrid = rb::rb_grid_1d((int64_t)N * M, 256);
  hipLaunchKernelGGL(skinny_combine_kernel, dim3(cgrid), dim3(256), 0,
                     stream, ws.slabs.data_ptr<float>(),
                     (uint16_t *)y.data_ptr(), M, N, ksplit);
  return y;
}
