// Skinny decode GEMM for CDNA4 (gfx950): y[M,N] = x[M,K] @ W[N,K]^T,
// M <= 32 (decode batch), bf16 in / bf16 out, fp32 accumulate.
//
// Why: batch-<=32 decode GEMMs are pure weight streams (W is ~99.9% of
// the traffic); hipBLASLt measured ~2 TB/s in the real decode loop
// (profiles/). The one thing that matters here is shaping the W read as
// long coalesced bursts and keeping enough of them in flight:
//
//  * grid = (N/64) x (K/256): each WG owns a [64 rows x 256 k] W tile
//    (32 KB) -> >= 1024 WGs on every llama shape, 3 WGs/CU by LDS, and
//    split-K slab traffic stays ~25% of the W stream (fp32 partials:
//    M*4 bytes per row-slice vs KT*2 of W).
//  * W tile is staged global->LDS with nontemporal 16 B/lane loads that
//    walk the tile row-major (256 B contiguous per row, rows adjacent in
//    the k-slice) — proper bursts, unlike direct MFMA A-fragment loads
//    whose 32 B/row at 8 KB stride waste the DRAM granule (that design
//    measured 1.5 TB/s; this file's history).
//  * x^T slice staged once in B-fragment-ready layout (tiny).
//  * main loop is pure LDS + MFMA: per 16-k step one ds_read_b128 (A,
//    row stride 272 B ≡ 4 mod 64 dwords -> conflict-free 16-lane groups,
//    same padding trick as attention_prefill.hip) + one ds_read_b128 (B)
//    + one v_mfma_f32_32x32x16_bf16.
//  * split-K partials: plain fp32 slab stores; the kernel boundary is
//    the release and skinny_combine reduces slabs -> bf16 y.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 128;        // 2 waves; wave owns 32 W rows
constexpr int NB = 64;            // W rows per WG
constexpr int KT = 256;           // k per WG
constexpr int MMAX = 32;
constexpr int W_STRIDE = KT * 2 + 16;  // LDS row bytes (+16: bank offset 4)

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(16))) float f32x16v;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4v;

__device__ __forceinline__ rb::bf16x8 nt_load8(const uint16_t *p) {
  union { u32x4v u; rb::bf16x8 v; } c;
  c.u = __builtin_nontemporal_load(reinterpret_cast<const u32x4v *>(p));
  return c.v;
}

// slabs layout: [nblk][kslice][NB][MMAX] f32
__global__ __launch_bounds__(BLOCK, 2) void skinny_gemm_kernel(
    const uint16_t *__restrict__ xp, const uint16_t *__restrict__ wp,
    float *__restrict__ slabs, int M, int N, int K) {
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;
  const int col = lane & 31;

  const int nblk = blockIdx.x;
  const int kslice = blockIdx.y;
  const int k0 = kslice * KT;
  const int ksplit = gridDim.y;

  __shared__ __attribute__((aligned(16))) char w_img[NB * W_STRIDE];
  __shared__ __attribute__((aligned(16))) uint16_t xfrag[KT * MMAX];

  // ---- stage W tile, coalesced + nontemporal (one-pass stream) ----------
  {
    constexpr int PER_ROW = KT / 8;         // 16B slots per row (16)
    constexpr int SLOTS = NB * PER_ROW;     // 2048
#pragma unroll
    for (int p = 0; p < SLOTS / BLOCK; ++p) {
      const int s = p * BLOCK + tid;
      const int r = s / PER_ROW;
      const int c8 = (s % PER_ROW) * 8;
      rb::bf16x8 v = nt_load8(
          wp + (int64_t)(nblk * NB + r) * K + k0 + c8);
      *reinterpret_cast<rb::bf16x8 *>(w_img + r * W_STRIDE + c8 * 2) = v;
    }
  }
  // ---- stage x^T slice in B-fragment-ready layout -----------------------
  {
    constexpr int SLOTS = (KT / 16) * 64;   // 512
    for (int s = tid; s < SLOTS; s += BLOCK) {
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

  // ---- LDS -> MFMA ------------------------------------------------------
  f32x16v acc = (f32x16v)(0.0f);
#pragma unroll
  for (int s = 0; s < KT / 16; ++s) {
    const bf16x8v a = *reinterpret_cast<const bf16x8v *>(
        w_img + (wid * 32 + col) * W_STRIDE + (s * 16 + hi * 8) * 2);
    const bf16x8v b = *reinterpret_cast<const bf16x8v *>(
        xfrag + s * 512 + (lane << 3));
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
  }

  // ---- partial tile -> slab (plain stores; boundary = release) ----------
  float *slab = slabs + (((int64_t)nblk * ksplit + kslice) * NB) * MMAX;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int n_local = wid * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
    slab[n_local * MMAX + col] = acc[r];
  }
}

// ---------------------------------------------------------------------------
// FP8 (OCP e4m3) weight-only variant: W stored as fp8 + per-output-channel
// f32 scales (absmax/448). Halves the decode weight stream — the entire
// cost of these GEMMs — while activations stay bf16 (no precision loss on
// x). Serves the container contract's MODEL_LOAD_IN_8BIT
// (reference examples/llama2-7b/server.yaml:9). Dequant happens on the
// LDS->register path with v_cvt_pk_f32_fp8 (gfx950-native, no LUT); the
// combine kernel applies the channel scale.
// ---------------------------------------------------------------------------

constexpr int W8_STRIDE = KT + 16;  // fp8 LDS row bytes (+16: bank offset 4)

__device__ __forceinline__ bf16x8v fp8x8_to_bf16(unsigned w0, unsigned w1) {
  auto a = __builtin_amdgcn_cvt_pk_f32_fp8(w0, false);
  auto b = __builtin_amdgcn_cvt_pk_f32_fp8(w0, true);
  auto c = __builtin_amdgcn_cvt_pk_f32_fp8(w1, false);
  auto d = __builtin_amdgcn_cvt_pk_f32_fp8(w1, true);
  union { unsigned short u[8]; bf16x8v v; } o;
  o.u[0] = rb::f32_to_bf16(a[0]); o.u[1] = rb::f32_to_bf16(a[1]);
  o.u[2] = rb::f32_to_bf16(b[0]); o.u[3] = rb::f32_to_bf16(b[1]);
  o.u[4] = rb::f32_to_bf16(c[0]); o.u[5] = rb::f32_to_bf16(c[1]);
  o.u[6] = rb::f32_to_bf16(d[0]); o.u[7] = rb::f32_to_bf16(d[1]);
  return o.v;
}

__global__ __launch_bounds__(BLOCK, 2) void skinny_gemm_fp8_kernel(
    const uint16_t *__restrict__ xp, const uint8_t *__restrict__ wp,
    float *__restrict__ slabs, int M, int N, int K) {
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;
  const int col = lane & 31;

  const int nblk = blockIdx.x;
  const int kslice = blockIdx.y;
  const int k0 = kslice * KT;
  const int ksplit = gridDim.y;

  __shared__ __attribute__((aligned(16))) uint8_t w_img[NB * W8_STRIDE];
  __shared__ __attribute__((aligned(16))) uint16_t xfrag[KT * MMAX];

  // stage fp8 W tile, coalesced (16 fp8 per 16 B slot)
  {
    constexpr int PER_ROW = KT / 16;
    constexpr int SLOTS = NB * PER_ROW;   // 1024
#pragma unroll
    for (int p = 0; p < SLOTS / BLOCK; ++p) {
      const int s = p * BLOCK + tid;
      const int r = s / PER_ROW;
      const int c16 = (s % PER_ROW) * 16;
      u32x4v v = __builtin_nontemporal_load(
          reinterpret_cast<const u32x4v *>(
              wp + (int64_t)(nblk * NB + r) * K + k0 + c16));
      *reinterpret_cast<u32x4v *>(w_img + r * W8_STRIDE + c16) = v;
    }
  }
  // stage x^T fragments (same as bf16 kernel)
  {
    constexpr int SLOTS = (KT / 16) * 64;
    for (int s = tid; s < SLOTS; s += BLOCK) {
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

  f32x16v acc = (f32x16v)(0.0f);
#pragma unroll
  for (int s = 0; s < KT / 16; ++s) {
    const uint8_t *wrow =
        w_img + (wid * 32 + col) * W8_STRIDE + s * 16 + hi * 8;
    const unsigned w0 = *reinterpret_cast<const unsigned *>(wrow);
    const unsigned w1 = *reinterpret_cast<const unsigned *>(wrow + 4);
    const bf16x8v a = fp8x8_to_bf16(w0, w1);
    const bf16x8v b = *reinterpret_cast<const bf16x8v *>(
        xfrag + s * 512 + (lane << 3));
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
  }

  float *slab = slabs + (((int64_t)nblk * ksplit + kslice) * NB) * MMAX;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int n_local = wid * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
    slab[n_local * MMAX + col] = acc[r];
  }
}

// Combine with per-channel scale (fp8 path).
__global__ void skinny_combine_scaled_kernel(
    const float *__restrict__ slabs, const float *__restrict__ scale,
    uint16_t *__restrict__ yp, int M, int N, int ksplit) {
  const int64_t total = (int64_t)N * M;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int n = (int)(i / M);
    const int m = (int)(i % M);
    const float *base = slabs +
        ((int64_t)(n / NB) * ksplit * NB + n % NB) * MMAX + m;
    float v = 0.0f;
    for (int ks = 0; ks < ksplit; ++ks) v += base[(int64_t)ks * NB * MMAX];
    yp[(int64_t)m * N + n] = rb::f32_to_bf16(v * scale[n]);
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

bool skinny_gemm_supported(int64_t M, int64_t N, int64_t K) {
  return M <= MMAX && N % NB == 0 && K % KT == 0;
}

at::Tensor skinny_gemm(at::Tensor x, at::Tensor w) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous(),
              "skinny_gemm: contiguous GPU tensors");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16, "skinny_gemm: bf16 only");
  const int M = (int)x.size(0), K = (int)x.size(1), N = (int)w.size(0);
  TORCH_CHECK((int)w.size(1) == K, "skinny_gemm: K mismatch");
  TORCH_CHECK(skinny_gemm_supported(M, N, K),
              "skinny_gemm: unsupported shape ", M, "x", N, "x", K);

  const int nblocks = N / NB;
  const int ksplit = K / KT;
  auto y = at::empty({M, N}, x.options());
  auto &ws = ws_for(x, (int64_t)nblocks * ksplit * NB * MMAX);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(skinny_gemm_kernel, dim3(nblocks, ksplit), dim3(BLOCK),
                     0, stream,
                     (const uint16_t *)x.data_ptr(),
                     (const uint16_t *)w.data_ptr(),
                     ws.slabs.data_ptr<float>(), M, N, K);
  const int cgrid = rb::rb_grid_1d((int64_t)N * M, 256);
  hipLaunchKernelGGL(skinny_combine_kernel, dim3(cgrid), dim3(256), 0,
                     stream, ws.slabs.data_ptr<float>(),
                     (uint16_t *)y.data_ptr(), M, N, K / KT);
  return y;
}

at::Tensor skinny_gemm_fp8(at::Tensor x, at::Tensor w8, at::Tensor scale) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w8.is_contiguous() &&
              scale.is_contiguous(), "skinny_gemm_fp8: contiguous");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "skinny_gemm_fp8: x bf16");
  TORCH_CHECK(scale.scalar_type() == at::kFloat, "skinny_gemm_fp8: scale f32");
  const int M = (int)x.size(0), K = (int)x.size(1), N = (int)w8.size(0);
  TORCH_CHECK((int)w8.size(1) == K && (int)scale.numel() == N,
              "skinny_gemm_fp8: shape");
  TORCH_CHECK(skinny_gemm_supported(M, N, K),
              "skinny_gemm_fp8: unsupported shape ", M, "x", N, "x", K);

  const int nblocks = N / NB;
  const int ksplit = K / KT;
  auto y = at::empty({M, N}, x.options());
  auto &ws = ws_for(x, (int64_t)nblocks * ksplit * NB * MMAX);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(skinny_gemm_fp8_kernel, dim3(nblocks, ksplit),
                     dim3(BLOCK), 0, stream,
                     (const uint16_t *)x.data_ptr(),
                     (const uint8_t *)w8.data_ptr(),
                     ws.slabs.data_ptr<float>(), M, N, K);
  const int cgrid = rb::rb_grid_1d((int64_t)N * M, 256);
  hipLaunchKernelGGL(skinny_combine_scaled_kernel, dim3(cgrid), dim3(256), 0,
                     stream, ws.slabs.data_ptr<float>(),
                     scale.data_ptr<float>(), (uint16_t *)y.data_ptr(),
                     M, N, ksplit);
  return y;
}
