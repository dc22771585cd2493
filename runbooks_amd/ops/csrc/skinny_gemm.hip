// Skinny decode GEMM for CDNA4 (gfx950): y[M,N] = x[M,K] @ W[N,K]^T,
// M <= 32 (decode batch), bf16 in / bf16 out, fp32 accumulate.
//
// Why: batch-<=32 decode GEMMs are pure weight streams (W is ~99.9% of
// the traffic) and hipBLASLt's tiles measured only 25-45% of HBM
// bandwidth on these shapes (profiles/). This kernel makes the W read
// the only significant traffic and shapes it as perfectly-coalesced
// 16 B/lane A-fragment loads:
//
//  * grid = (N/128 n-blocks) x KSPLIT k-slices  -> >= 2 WGs/CU on every
//    llama/falcon shape (XCD-aware via plain block order; traffic is
//    uniform so placement only matters for L2 reuse of x).
//  * WG = 4 waves; wave owns 32 rows of W. Per 16-k step it issues ONE
//    global_load_dwordx4 (its A fragment, 32 rows x 16 k) and ONE
//    ds_read_b128 (x^T fragment staged in B-fragment-ready layout), one
//    v_mfma_f32_32x32x16_bf16. HBM-bound by construction; nt loads keep
//    the one-pass weight stream out of L2 (MI355X_MICROARCH.md
//    nt-weights row).
//  * split-K partials land in fp32 slabs; an arrival ticket elects the
//    last WG per n-block to combine slabs -> bf16 y (the guide's
//    splitk-seam/publish-large recipe: release fence + vmcnt(0) before
//    the ticket, acquire fence after winning it).
//
// mfma_f32_32x32x16_bf16 layouts as in attention_prefill.hip.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;        // 4 waves
constexpr int NB = 128;           // W rows per WG (32 per wave)
constexpr int MMAX = 32;

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(16))) float f32x16v;

// y = x @ W^T. Workspace: partial [KSPLIT][N][MMAX] f32 laid out as
// [nblk][KSPLIT][NB][MMAX]; tickets: one int per n-block (pre-zeroed).
__global__ __launch_bounds__(BLOCK, 4) void skinny_gemm_kernel(
    const uint16_t *__restrict__ xp, const uint16_t *__restrict__ wp,
    uint16_t *__restrict__ yp, float *__restrict__ slabs,
    int *__restrict__ tickets, int M, int N, int K, int ksplit) {
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;
  const int col = lane & 31;

  const int nblk = blockIdx.x;          // which 128-row block of W
  const int kslice = blockIdx.y;        // which K slice
  const int n0 = nblk * NB + wid * 32;  // this wave's first W row
  const int kper = K / ksplit;          // k elems this WG contracts
  const int k0 = kslice * kper;

  // ---- stage x^T slice in B-fragment-ready layout -----------------------
  // frag_buf[(k_step)][lane][e]: element x[m = lane&31][k0 + k_step*16 +
  // (lane>>5)*8 + e]; 16 B per lane per ds_read_b128 in the main loop.
  extern __shared__ __attribute__((aligned(16))) uint16_t xfrag[];
  {
    // one 8-elem slot per iteration: slot s covers lane l of 16-k step ks,
    // a contiguous 8-k run of x row (l&31) -> one bf16x8 load + store.
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

  // ---- main loop: one A load + one LDS read + one MFMA per 16 k ---------
  f32x16v acc = (f32x16v)(0.0f);
  const int steps = kper / 16;
  for (int s = 0; s < steps; ++s) {
    // A fragment: W[n0 + col][k0 + s*16 + hi*8 .. +8), nt (one-pass stream)
    const uint16_t *wrow = wp + (int64_t)(n0 + col) * K + k0 + s * 16 + hi * 8;
    typedef __attribute__((ext_vector_type(4))) unsigned int u32x4v;
    bf16x8v a;
    {
      union { u32x4v u; bf16x8v v; } c;
      c.u = __builtin_nontemporal_load(
          reinterpret_cast<const u32x4v *>(wrow));
      a = c.v;
    }
    const bf16x8v b = *reinterpret_cast<const bf16x8v *>(
        xfrag + s * 512 + (lane << 3));
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
  }

  // ---- publish this slice's partial tile --------------------------------
  // slab tile [NB][MMAX] f32; C reg r -> n_local = wid*32 + (r&3)+8*(r>>2)
  // +4*hi, m = col.
  float *slab = slabs + (((int64_t)nblk * ksplit + kslice) * NB) * MMAX;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int n_local = wid * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
    slab[n_local * MMAX + col] = acc[r];
  }

  if (ksplit == 1) {
    // single slice: convert directly, no ticket round-trip
    __syncthreads();
    float *base = slabs + ((int64_t)nblk * ksplit * NB) * MMAX;
    for (int i = tid; i < NB * M; i += BLOCK) {
      const int n_local = i / M;
      const int m = i % M;
      yp[(int64_t)m * N + nblk * NB + n_local] =
          rb::f32_to_bf16(base[n_local * MMAX + m]);
    }
    return;
  }

  // release the slab, then take a ticket (guide §6 G16 / splitk-seam)
  __syncthreads();
  __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __shared__ int my_ticket;
  if (tid == 0)
    my_ticket = __hip_atomic_fetch_add(&tickets[nblk], 1,
                                       __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
  __syncthreads();
  if (my_ticket != ksplit - 1)
    return;  // not the last arriver

  // ---- last arriver: combine all slices of this n-block -> y ------------
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  if (tid == 0)
    tickets[nblk] = 0;  // re-armed for the next launch (graph replay)
  float *base = slabs + ((int64_t)nblk * ksplit * NB) * MMAX;
  for (int i = tid; i < NB * M; i += BLOCK) {
    const int n_local = i / M;
    const int m = i % M;
    float v = 0.0f;
    for (int ks = 0; ks < ksplit; ++ks)
      v += base[((int64_t)ks * NB + n_local) * MMAX + m];
    yp[(int64_t)m * N + nblk * NB + n_local] = rb::f32_to_bf16(v);
  }
}

}  // namespace

// Workspace cache: slabs + tickets sized for the largest shape seen; the
// decode graph replays with stable pointers.
namespace {
struct SkinnyWorkspace {
  at::Tensor slabs;
  at::Tensor tickets;
};
SkinnyWorkspace &ws_for(const at::Tensor &ref, int nblocks, int ksplit) {
  static SkinnyWorkspace ws;
  const int64_t need = (int64_t)nblocks * ksplit * NB * MMAX;
  if (!ws.slabs.defined() || ws.slabs.numel() < need ||
      ws.slabs.device() != ref.device()) {
    ws.slabs = at::empty({need}, ref.options().dtype(at::kFloat));
    ws.tickets = at::zeros({4096}, ref.options().dtype(at::kInt));
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
  // fill ~2 WGs per CU; k slices must keep kper % 16 == 0
  int ksplit = 1;
  while (nblocks * ksplit < 512 && ksplit < 16 &&
         (K / (ksplit * 2)) % 16 == 0 && K / (ksplit * 2) >= 64)
    ksplit *= 2;

  auto y = at::empty({M, N}, x.options());
  auto &ws = ws_for(x, nblocks, ksplit);
  auto stream = at::hip::getCurrentHIPStream();
  const size_t shmem = (size_t)(K / ksplit) * MMAX * sizeof(uint16_t);
  TORCH_CHECK(shmem <= 160 * 1024, "skinny_gemm: K/ksplit too large");
  hipLaunchKernelGGL(skinny_gemm_kernel, dim3(nblocks, ksplit), dim3(BLOCK),
                     shmem, stream,
                     (const uint16_t *)x.data_ptr(),
                     (const uint16_t *)w.data_ptr(),
                     (uint16_t *)y.data_ptr(), ws.slabs.data_ptr<float>(),
                     ws.tickets.data_ptr<int>(), M, N, K, ksplit);
  return y;
}
