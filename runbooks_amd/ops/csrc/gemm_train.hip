// EXPERIMENTAL: 256^2 8-phase bf16 GEMM for CDNA4 (gfx950).
//
// C[M,N] = A[M,K] @ B[N,K]^T (both row-major, contracted over the last
// dim — the nn.Linear forward shape). Target: the training-step
// projection GEMMs (M=2048), where hipBLASLt measured 860-1140 TF/s
// in-situ (profiles/train_*.csv) while this template's verified ceiling
// is ~1320-1340 TF/s on random operands
// (cdna_hip_programming.md §5 "The 256^2 8-phase template").
//
// Status: drafted offline at the end of round 1 (no GPU budget left to
// validate); NOT wired into any model path. Gated behind
// RB_EXPERIMENTAL: tests skip unless it is set. Round 2: run
// `RB_EXPERIMENTAL=1 pytest tests/test_gpu_ops.py -k train_gemm` first,
// then benchmarks/kernels.py.
//
// Grid fill: when (M/256)*(N/256) < 256 the host splits K (fp32 slabs
// + a combine kernel, the skinny_gemm.hip pattern) — M=2048 N=4096 runs
// as 128 tiles x ksplit 2 = 256 WGs. The whole index dataflow (staging
// swizzle, fragment maps, phase accumulation, split-K slabs + combine,
// epilogue) is verified by the CPU simulation in
// tests/test_gemm_sim_cpu.py; keep the two in sync.
//
// Template geometry (guide table):
//   tile BM x BN = 256 x 256, BK = 64, 8 waves (2M x 4N), 512 threads
//   LDS 128 KiB = 2 dbuf x 2 ops(A,B) x 2 halves x 128 rows x 64 k x 2B
//   per-wave output 128 x 64 = acc[8][4] fragments of 16x16 (f32x4)
//   mfma_f32_16x16x32_bf16; 64 MFMA / K-tile / wave; 16 per phase
//   glds: __builtin_amdgcn_global_load_lds width 16; 2 per half-tile
//   LDS swizzle st_16x32: byte ^= ((byte>>9)&1)<<5 (within 1 KiB subtile)
//   raw s_barrier + lgkmcnt in the loop; vmcnt(6) at phases 4 and 8 only
//
// mfma_f32_16x16x32_bf16 layouts (cdna_hip_programming.md §3):
//   A[i][k]: i = lane&15, k = (lane>>4)*8 + e   (e = 0..7)
//   B[k][j]: j = lane&15, k = (lane>>4)*8 + e
//   C[i][j]: j = lane&15, i = (lane>>4)*4 + reg (reg = 0..3)

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int THREADS = 512;             // 8 waves, 2(M) x 4(N)
constexpr int HALF_BYTES = 128 * BK * 2; // one half-tile in LDS (16 KiB)
constexpr int OP_BYTES = 2 * HALF_BYTES; // A or B full tile (32 KiB)
constexpr int DB_BYTES = 2 * OP_BYTES;   // A+B for one K-tile (64 KiB)

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(4))) float f32x4v;

// st_16x32 swizzle on a byte offset within a 16 KiB half-tile image.
__device__ __forceinline__ int swz(int byte) {
  return byte ^ (((byte >> 9) & 1) << 5);
}

// One global_load_lds pass: stage a [128 x 64] bf16 half-tile (row-major,
// row stride ld elements) into LDS at image offset lds_off, pre-swizzling
// the SOURCE address so the (lane-linear) LDS image is the swizzled
// layout. Each glds writes wave-uniform-base + lane*16: one instruction
// per WAVE covers 1 KiB, 8 waves x 2 passes = 16 KiB.
__device__ __forceinline__ void stage_half(char *lds, const uint16_t *src,
                                           int ld, uint32_t lds_off, int tid,
                                           int pass) {
  const int wid = tid >> 6;
  // LDS byte this lane writes (linear within the image):
  const int b = pass * (THREADS * 16) + tid * 16;
  const int q = swz(b);                  // logical byte it must hold
  const int row = q >> 7;                // 128 B per logical row
  const int col2 = q & 127;              // byte within row
  const uint16_t *g = src + (int64_t)row * ld + (col2 >> 1);
  // wave-uniform LDS base for THIS wave's 1 KiB segment
  char *dst = lds + lds_off + pass * (THREADS * 16) + wid * (RB_WAVE * 16);
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) uint32_t *)(const void *)(g),
      (__attribute__((address_space(3))) uint32_t *)(void *)(dst),
      16, 0, 0);
}

__device__ __forceinline__ bf16x8v lds_read_frag(const char *lds,
                                                 int logical_byte) {
  return *reinterpret_cast<const bf16x8v *>(lds + swz(logical_byte));
}

// cp: bf16 output when gridDim.z == 1; otherwise fp32 partial slabs
// [tile][kslice][BM][BN] (combined by gemm_nt_combine_kernel — the
// kernel boundary is the release, as in skinny_gemm.hip).
__global__ __launch_bounds__(THREADS, 1) void gemm_nt_8phase_kernel(
    const uint16_t *__restrict__ ap, const uint16_t *__restrict__ bp,
    uint16_t *__restrict__ cp, float *__restrict__ slabs,
    int M, int N, int K) {
  extern __shared__ __attribute__((aligned(16))) char lds[];

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int wm = wid >> 2;               // 0..1
  const int wn = wid & 3;                // 0..3

  // XCD-aware bijective workgroup remap (guide: XCD swizzle)
  int nwg = gridDim.x * gridDim.y;
  int orig = blockIdx.y * gridDim.x + blockIdx.x;
  int q8 = nwg / 8, r8 = nwg % 8;
  int xcd = orig % 8, pos = orig / 8;
  int wg = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + pos;
  const int ntiles_n = N / BN;
  const int tile_m = wg / ntiles_n;
  const int tile_n = wg % ntiles_n;

  const int ksplit = gridDim.z;
  const int kslice = blockIdx.z;
  const int kper = K / ksplit;             // this slice's contraction depth
  const uint16_t *a_tile = ap + (int64_t)tile_m * BM * K + kslice * kper;
  const uint16_t *b_tile = bp + (int64_t)tile_n * BN * K + kslice * kper;

  // LDS layout: [db][op][half] images of 16 KiB each
  auto lds_img = [&](int db, int op, int half) -> uint32_t {
    return (uint32_t)(db * DB_BYTES + op * OP_BYTES + half * HALF_BYTES);
  };

  // ---- prologue: stage K-tiles 0 and half of 1 --------------------------
  // order: A0h0 A0h1 B0h0 B0h1 | A1h0 A1h1 B1h0 (7 half-tiles, 14 glds)
  const int ktiles = kper / BK;
  {
    for (int p = 0; p < 2; ++p) stage_half(lds, a_tile + 0 * K + 0, K,
                                           lds_img(0, 0, 0), tid, p);
    for (int p = 0; p < 2; ++p) stage_half(lds, a_tile + 128 * K + 0, K,
                                           lds_img(0, 0, 1), tid, p);
    for (int p = 0; p < 2; ++p) stage_half(lds, b_tile + 0 * K + 0, K,
                                           lds_img(0, 1, 0), tid, p);
    for (int p = 0; p < 2; ++p) stage_half(lds, b_tile + 128 * K + 0, K,
                                           lds_img(0, 1, 1), tid, p);
    if (ktiles > 1) {
      for (int p = 0; p < 2; ++p) stage_half(lds, a_tile + 0 * K + BK, K,
                                             lds_img(1, 0, 0), tid, p);
      for (int p = 0; p < 2; ++p) stage_half(lds, a_tile + 128 * K + BK, K,
                                             lds_img(1, 0, 1), tid, p);
      for (int p = 0; p < 2; ++p) stage_half(lds, b_tile + 0 * K + BK, K,
                                             lds_img(1, 1, 0), tid, p);
    }
    // K-tile 0 fully landed (glds are FIFO per wave): 6 outstanding
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  f32x4v acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4v)(0.0f);

  // A fragment logical byte: half = wm (wave's 128 rows live in one half);
  // row_in_half = fm*16 + (lane&15); k = ks*32 + (lane>>4)*8
  auto a_byte = [&](int fm, int ks) {
    return ((fm * 16 + (lane & 15)) * BK + ks * 32 + ((lane >> 4) * 8)) * 2;
  };
  // B fragment: n = wn*64 + fn*16 + (lane&15); half = n>>7; row = n&127
  auto b_half = [&](int fn) { return (wn * 64 + fn * 16) >> 7; };
  auto b_byte = [&](int fn, int ks) {
    return (((wn * 64 + fn * 16 + (lane & 15)) & 127) * BK + ks * 32 +
            ((lane >> 4) * 8)) * 2;
  };

  bf16x8v af[4][2];   // one fm-half (4 fm) x 2 ks
  bf16x8v bf[2][2][2];  // both fn-halves (2 fn each) x 2 ks

  // next half-tile to prefetch, in the fixed order A.h0 A.h1 B.h0 B.h1
  int pf_kt = 1, pf_slot = 3;  // prologue already staged through (1, B, h0)

  auto prefetch_one = [&](int) {
    if (pf_kt < ktiles) {
      const int op = pf_slot >> 1, half = pf_slot & 1;
      const uint16_t *src = (op == 0 ? a_tile : b_tile) +
          (int64_t)(half * 128) * K + pf_kt * BK;
      const uint32_t img = lds_img(pf_kt & 1, op, half);
      stage_half(lds, src, K, img, tid, 0);
      stage_half(lds, src, K, img, tid, 1);
      if (++pf_slot == 4) { pf_slot = 0; ++pf_kt; }
    }
  };

  for (int kt = 0; kt < ktiles; ++kt) {
    const int db = kt & 1;
    const char *A = lds + lds_img(db, 0, wm);
    const char *B0 = lds + lds_img(db, 1, b_half(0));
    const char *B1 = lds + lds_img(db, 1, b_half(2));
    // NOTE: fn 0,1 share a half iff wn*64+16 stays within it — with
    // fn*16 <= 48 and wn*64 base, fn 0..3 of one wave span at most one
    // 128-row boundary only when wn == 1 (64..127) or wn == 2 — in fact
    // wn*64 + 63 < 128 for wn<2 and >= 128 for wn>=2, so all four fn of
    // a wave live in ONE half: half = wn >> 1.
    const char *B = lds + lds_img(db, 1, wn >> 1);

    // ---- phase 1: A(fmh0) 8 reads + B(fnh0) 4 reads, 16 mfma ------------
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        af[fm][ks] = lds_read_frag(A, a_byte(fm, ks));
#pragma unroll
    for (int fn = 0; fn < 2; ++fn)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        bf[0][fn][ks] = lds_read_frag(B, b_byte(fn, ks));
    prefetch_one(0);
    asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fm][ks], bf[0][fn][ks], acc[fm][fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---- phase 2: B(fnh1) 4 reads, reuse A(fmh0) ------------------------
#pragma unroll
    for (int fn = 0; fn < 2; ++fn)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        bf[1][fn][ks] = lds_read_frag(B, b_byte(2 + fn, ks));
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[fm][2 + fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fm][ks], bf[1][fn][ks], acc[fm][2 + fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---- phase 3: A(fmh1) 8 reads, reuse B(fnh1) ------------------------
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        af[fm][ks] = lds_read_frag(A, a_byte(4 + fm, ks));
    asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[4 + fm][2 + fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fm][ks], bf[1][fn][ks], acc[4 + fm][2 + fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---- phase 4: reuse A(fmh1) + B(fnh0); vmcnt gate for next K-tile ---
    // The three stages of tile kt+2 land in db(kt)'s LDS image, which
    // phases 1-3 of THIS tile still read — issuing them any earlier is a
    // data race (r1 shipped them in phases 2/3: wrong outputs on every
    // shape big enough for the glds to land before phase 3's reads,
    // GPUTEST r9 bisect). Phase 4 reads no LDS (registers only) and
    // starts after the phase-3-end barrier, so every wave's reads of
    // db(kt) are complete here.
    prefetch_one(1);
    prefetch_one(2);
    prefetch_one(3);
    if (kt + 1 < ktiles) {
      // gate the NEXT K-tile's data: leave in flight only half-tiles
      // belonging beyond kt+1 (3 halves = 6 glds in steady state; at the
      // tail nothing is ahead, so drain fully)
      if (kt + 2 < ktiles)
        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[4 + fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fm][ks], bf[0][fn][ks], acc[4 + fm][fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  }

  // ---- epilogue: acc -> C (bf16) or fp32 slab ---------------------------
  // C[i=(lane>>4)*4+e][j=lane&15] per 16x16 fragment.
  if (ksplit == 1) {
    const int crow0 = tile_m * BM + wm * 128;
    const int ccol0 = tile_n * BN + wn * 64;
#pragma unroll
    for (int fm = 0; fm < 8; ++fm)
#pragma unroll
      for (int fn = 0; fn < 4; ++fn) {
        const int r0 = crow0 + fm * 16 + ((lane >> 4) * 4);
        const int c = ccol0 + fn * 16 + (lane & 15);
#pragma unroll
        for (int e = 0; e < 4; ++e)
          cp[(int64_t)(r0 + e) * N + c] = rb::f32_to_bf16(acc[fm][fn][e]);
      }
  } else {
    float *slab = slabs +
        (((int64_t)wg * ksplit + kslice) * BM) * BN;  // [tile][slice] image
    const int srow0 = wm * 128;
    const int scol0 = wn * 64;
#pragma unroll
    for (int fm = 0; fm < 8; ++fm)
#pragma unroll
      for (int fn = 0; fn < 4; ++fn) {
        const int r0 = srow0 + fm * 16 + ((lane >> 4) * 4);
        const int c = scol0 + fn * 16 + (lane & 15);
#pragma unroll
        for (int e = 0; e < 4; ++e)
          slab[(int64_t)(r0 + e) * BN + c] = acc[fm][fn][e];
      }
  }
}

// Combine ksplit fp32 slabs -> bf16 C. wg/tile mapping mirrors the main
// kernel's (post-XCD-remap wg index owns slab block wg).
__global__ void gemm_nt_combine_kernel(const float *__restrict__ slabs,
                                       uint16_t *__restrict__ cp,
                                       int M, int N, int ksplit) {
  const int ntiles_n = N / BN;
  const int64_t total = (int64_t)M * N;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int r = (int)(i / N), c = (int)(i % N);
    const int wg = (r / BM) * ntiles_n + (c / BN);
    const float *base = slabs + (((int64_t)wg * ksplit) * BM) * BN +
        (int64_t)(r % BM) * BN + (c % BN);
    float v = 0.0f;
    for (int ks = 0; ks < ksplit; ++ks) v += base[(int64_t)ks * BM * BN];
    cp[i] = rb::f32_to_bf16(v);
  }
}

}  // namespace

at::Tensor train_gemm_nt(at::Tensor a, at::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && b.is_contiguous(),
              "train_gemm_nt: contiguous GPU tensors");
  TORCH_CHECK(a.scalar_type() == at::kBFloat16 &&
              b.scalar_type() == at::kBFloat16, "train_gemm_nt: bf16");
  const int M = (int)a.size(0), K = (int)a.size(1), N = (int)b.size(0);
  TORCH_CHECK((int)b.size(1) == K, "train_gemm_nt: K mismatch");
  TORCH_CHECK(M % BM == 0 && N % BN == 0 && K % (2 * BK) == 0,
              "train_gemm_nt: M%256, N%256, K%128 required (experimental)");
  // split K until the grid covers the chip (fixes the M=2048 N=4096
  // underfill: 128 tiles -> ksplit 2 -> 256 WGs)
  const int tiles = (M / BM) * (N / BN);
  int ksplit = 1;
  while (tiles * ksplit < 256 && ksplit < 8 &&
         (K / (ksplit * 2)) % (2 * BK) == 0)
    ksplit *= 2;

  auto c = at::empty({M, N}, a.options());
  auto stream = at::hip::getCurrentHIPStream();
  const dim3 grid(M / BM, N / BN, ksplit);
  constexpr size_t shmem = 2 * DB_BYTES;   // 128 KiB
  static bool cfg_done = false;
  if (!cfg_done) {
    hipFuncSetAttribute(
        reinterpret_cast<const void *>(&gemm_nt_8phase_kernel),
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)shmem);
    cfg_done = true;
  }
  float *slabs = nullptr;
  at::Tensor ws;
  if (ksplit > 1) {
    static at::Tensor ws_cache;
    const int64_t need = (int64_t)tiles * ksplit * BM * BN;
    if (!ws_cache.defined() || ws_cache.numel() < need ||
        ws_cache.device() != a.device())
      ws_cache = at::empty({need}, a.options().dtype(at::kFloat));
    ws = ws_cache;
    slabs = ws.data_ptr<float>();
  }
  hipLaunchKernelGGL(gemm_nt_8phase_kernel, grid, dim3(THREADS), shmem,
                     stream, (const uint16_t *)a.data_ptr(),
                     (const uint16_t *)b.data_ptr(),
                     (uint16_t *)c.data_ptr(), slabs, M, N, K);
  if (ksplit > 1) {
    const int cgrid = rb::rb_grid_1d((int64_t)M * N, 256);
    hipLaunchKernelGGL(gemm_nt_combine_kernel, dim3(cgrid), dim3(256), 0,
                       stream, slabs, (uint16_t *)c.data_ptr(), M, N,
                       ksplit);
  }
  return c;
}

// ---------------------------------------------------------------------------
// Layout probe for mfma_f32_16x16x32_bf16 (the 32x32 probe lives in
// attention_prefill.hip): one MFMA as a plain matmul so a wrong fragment
// map fails here first.
// a: [16, 32] bf16 (A[i][k]); b: [32, 16] bf16 (B[k][j]) -> c: [16, 16] f32
// ---------------------------------------------------------------------------
namespace {
__global__ void mfma16_probe_kernel(const uint16_t *__restrict__ a,
                                    const uint16_t *__restrict__ b,
                                    float *__restrict__ c) {
  const int lane = threadIdx.x & 63;
  const int i = lane & 15;
  const int kq = lane >> 4;  // 0..3
  union { unsigned short u[8]; bf16x8v v; } af, bfv;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    af.u[e] = a[i * 32 + kq * 8 + e];      // A[i][k]
    bfv.u[e] = b[(kq * 8 + e) * 16 + i];   // B[k][j=i]
  }
  f32x4v acc = (f32x4v)(0.0f);
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af.v, bfv.v, acc, 0, 0, 0);
#pragma unroll
  for (int e = 0; e < 4; ++e)
    c[(kq * 4 + e) * 16 + i] = acc[e];     // C[row][j]
}
}  // namespace

at::Tensor mfma_probe_16x16x32(at::Tensor a, at::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16);
  TORCH_CHECK(a.sizes() == at::IntArrayRef({16, 32}) &&
              b.sizes() == at::IntArrayRef({32, 16}), "probe shapes");
  auto c = at::zeros({16, 16}, a.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma16_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const uint16_t *)a.contiguous().data_ptr(),
                     (const uint16_t *)b.contiguous().data_ptr(),
                     c.data_ptr<float>());
  return c;
}
