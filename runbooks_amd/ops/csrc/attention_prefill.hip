// Flash-attention prefill (causal, GQA/MQA, bf16) for CDNA4 (gfx950).
//
// MFMA-native design following the CDNA4 attention recipe:
//  * workgroup = 8 waves; each wave owns 32 query rows (WG tile = 256).
//  * KV tiles of 64 keys staged to LDS; K row-major, V TRANSPOSED
//    ([d][key]) so both QK^T and PV read 16 B/lane fragments. Rows padded
//    +16 B so the b128 column-slice reads are bank-conflict-free
//    (stride/4 mod 64 cycles through distinct banks; no XOR swizzle).
//  * swapped QK^T: mfma(A=K, B=Q) so the C layout has qrow = lane&31 —
//    the whole softmax row lives on a lane pair (l, l^32); row max is a
//    16-reg in-lane reduce + one shfl_xor(32).
//  * online softmax in registers; P repacked to PV A/B fragments with
//    v_cvt (compiler) + v_permlane32_swap (half exchange).
//  * PV: mfma(A=V^T, B=P^T) accumulating O^T in 16-reg tiles.
//
// mfma_f32_32x32x16_bf16 layouts (cdna_hip_programming.md §3):
//   A[i][k]: i = lane&31, k = (lane>>5)*8 + e (e = 0..7)
//   C[i][j]: j = lane&31, i = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
//
// q,k,v,out: [B, S, H, DH] bf16 token-major. DH in {64, 128}.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 512;   // 8 waves
constexpr int KVB = 64;      // keys per LDS tile

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(16))) float f32x16v;

RB_DEV unsigned pack_bf16(float lo, float hi) {
  union { __bf16 b; unsigned short u; } a, b;
  a.b = (__bf16)lo;
  b.b = (__bf16)hi;
  return ((unsigned)b.u << 16) | a.u;
}

RB_DEV bf16x8v frag_from_words(unsigned w0, unsigned w1, unsigned w2, unsigned w3) {
  union { unsigned u[4]; bf16x8v v; } c;
  c.u[0] = w0; c.u[1] = w1; c.u[2] = w2; c.u[3] = w3;
  return c.v;
}

template <int DH>
__global__ __launch_bounds__(BLOCK, 2) void flash_prefill_kernel(
    const uint16_t *__restrict__ qp, const uint16_t *__restrict__ kp,
    const uint16_t *__restrict__ vp, uint16_t *__restrict__ op,
    float *__restrict__ lsep,  // [B, Hq, S] log-sum-exp for training bwd
    int B, int S, int Hq, int Hkv, float scale) {
  constexpr int KSTEPS = DH / 16;       // QK^T contraction steps
  constexpr int DTILES = DH / 32;       // O column tiles
  constexpr int K_STRIDE = DH * 2 + 16; // K row bytes (+16 pad)
  constexpr int V_STRIDE = KVB * 2 + 16;  // V^T row bytes (+16 pad)

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char *k_img = smem;                       // [KVB][K_STRIDE]
  char *v_img = smem + KVB * K_STRIDE;      // [DH][V_STRIDE]

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;
  const int col = lane & 31;                // qrow within the wave tile

  const int q0 = blockIdx.x * 256;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int h_kv = h / (Hq / Hkv);
  const int qrow = q0 + wid * 32 + col;     // this lane's query row
  const bool q_valid = qrow < S;

  // ---- Q fragments (B operand of swapped QK^T), pre-scaled --------------
  bf16x8v qf[KSTEPS];
  {
    const uint16_t *qrow_p =
        qp + ((int64_t)(b * S + (q_valid ? qrow : 0)) * Hq + h) * DH;
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      const uint16_t *src = qrow_p + ks * 16 + hi * 8;
      float f[8];
      rb::VIO<uint16_t>::load(src, f);
      union { unsigned short u[8]; bf16x8v v; } c;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        c.u[e] = rb::f32_to_bf16(q_valid ? f[e] * scale : 0.0f);
      qf[ks] = c.v;
    }
  }

  f32x16v acc_o[DTILES];
#pragma unroll
  for (int dt = 0; dt < DTILES; ++dt) acc_o[dt] = (f32x16v)(0.0f);
  float m_run = -INFINITY;
  float l_run = 0.0f;

  const int kv_end = min(S, q0 + 256);      // causal upper bound for this WG
  const int n_tiles = (kv_end + KVB - 1) / KVB;
  const int wave_kmax = q0 + wid * 32 + 31; // last key this wave can see

  for (int kt = 0; kt < n_tiles; ++kt) {
    const int kbase = kt * KVB;

    // ---- stage K tile: [key][d] rows, each thread 16 elems ---------------
    {
      const int per_pass = BLOCK * 8;       // elems per pass
      constexpr int total = KVB * DH;
#pragma unroll
      for (int p = 0; p < total / (BLOCK * 8); ++p) {
        const int idx = p * per_pass + tid * 8;
        const int key = idx / DH;
        const int d0 = idx % DH;
        const int gk = kbase + key;
        rb::bf16x8 val;
        if (gk < S) {
          val = *reinterpret_cast<const rb::bf16x8 *>(
              kp + ((int64_t)(b * S + gk) * Hkv + h_kv) * DH + d0);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) val.v[e] = 0;
        }
        *reinterpret_cast<rb::bf16x8 *>(k_img + key * K_STRIDE + d0 * 2) = val;
      }
    }
    // ---- stage V tile transposed: thread owns column d, KPG keys ---------
    {
      constexpr int GRPS = BLOCK / DH;      // thread groups over keys
      constexpr int KPG = KVB / GRPS;       // keys per group (16 @DH=128)
      const int d = tid % DH;
      const int kg0 = (tid / DH) * KPG;
      uint16_t tmp[KPG];
#pragma unroll
      for (int e = 0; e < KPG; ++e) {
        const int gk = kbase + kg0 + e;
        tmp[e] = (gk < S)
            ? vp[((int64_t)(b * S + gk) * Hkv + h_kv) * DH + d]
            : (uint16_t)0;
      }
#pragma unroll
      for (int c8 = 0; c8 < KPG / 8; ++c8)
        *reinterpret_cast<rb::bf16x8 *>(v_img + d * V_STRIDE + (kg0 + c8 * 8) * 2) =
            *reinterpret_cast<rb::bf16x8 *>(&tmp[c8 * 8]);
    }
    __syncthreads();

    if (kbase <= wave_kmax) {
      // two 32-key sub-tiles per LDS tile
#pragma unroll
      for (int it = 0; it < 2; ++it) {
        // ---- QK^T^T: scores[key][qrow] ----------------------------------
        f32x16v s = (f32x16v)(0.0f);
#pragma unroll
        for (int ks = 0; ks < KSTEPS; ++ks) {
          const bf16x8v kf = *reinterpret_cast<const bf16x8v *>(
              k_img + (it * 32 + col) * K_STRIDE + (ks * 16 + hi * 8) * 2);
          s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[ks], s, 0, 0, 0);
        }

        // ---- mask + online softmax --------------------------------------
        float p[16];
        float mx = -INFINITY;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int key = kbase + it * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          p[r] = (key <= qrow) ? s[r] : -INFINITY;
          mx = fmaxf(mx, p[r]);
        }
        mx = fmaxf(mx, __shfl_xor(mx, 32, 64));     // full row max
        const float mn = fmaxf(m_run, mx);
        if (mn != -INFINITY) {
          const float alpha = (m_run == -INFINITY) ? 0.0f : __expf(m_run - mn);
          float psum = 0.0f;
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            p[r] = (p[r] == -INFINITY) ? 0.0f : __expf(p[r] - mn);
            psum += p[r];
          }
          l_run = l_run * alpha + psum;
          m_run = mn;
          if (alpha != 1.0f) {
#pragma unroll
            for (int dt = 0; dt < DTILES; ++dt)
#pragma unroll
              for (int r = 0; r < 16; ++r) acc_o[dt][r] *= alpha;
          }

          // ---- P -> bf16 B-fragments via permlane half-exchange ---------
#pragma unroll
          for (int kslot = 0; kslot < 2; ++kslot) {
            const int r0 = kslot * 8;
            unsigned pk0 = pack_bf16(p[r0 + 0], p[r0 + 1]);
            unsigned pk1 = pack_bf16(p[r0 + 2], p[r0 + 3]);
            unsigned pk2 = pack_bf16(p[r0 + 4], p[r0 + 5]);
            unsigned pk3 = pack_bf16(p[r0 + 6], p[r0 + 7]);
            auto r02 = __builtin_amdgcn_permlane32_swap(pk0, pk2, false, false);
            auto r13 = __builtin_amdgcn_permlane32_swap(pk1, pk3, false, false);
            const bf16x8v pf = frag_from_words(r02[0], r13[0], r02[1], r13[1]);
            const int koff = (it * 32 + kslot * 16 + hi * 8) * 2;
#pragma unroll
            for (int dt = 0; dt < DTILES; ++dt) {
              const bf16x8v vf = *reinterpret_cast<const bf16x8v *>(
                  v_img + (dt * 32 + col) * V_STRIDE + koff);
              acc_o[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                  vf, pf, acc_o[dt], 0, 0, 0);
            }
          }
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: O^T regs -> out[b, qrow, h, :] ---------------------------
  if (q_valid) {
    const float l_full = l_run + __shfl_xor(l_run, 32, 64);
    const float inv_l = (l_full > 0.0f) ? 1.0f / l_full : 0.0f;
    if (lsep != nullptr && hi == 0)
      lsep[((int64_t)b * Hq + h) * S + qrow] =
          (l_full > 0.0f) ? m_run + __logf(l_full) : -INFINITY;
    uint16_t *orow = op + ((int64_t)(b * S + qrow) * Hq + h) * DH;
#pragma unroll
    for (int dt = 0; dt < DTILES; ++dt) {
#pragma unroll
      for (int rq = 0; rq < 4; ++rq) {
        const int d0 = dt * 32 + 8 * rq + 4 * hi;
        uint16_t w[4];
#pragma unroll
        for (int e = 0; e < 4; ++e)
          w[e] = rb::f32_to_bf16(acc_o[dt][rq * 4 + e] * inv_l);
        *reinterpret_cast<uint2 *>(orow + d0) =
            *reinterpret_cast<const uint2 *>(w);
      }
    }
  }
}

}  // namespace

static at::Tensor flash_fwd_impl(at::Tensor q, at::Tensor k, at::Tensor v,
                                 double scale, float *lsep) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous() &&
                  v.is_contiguous(), "flash_prefill: contiguous GPU tensors");
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "flash_prefill: bf16 only");
  TORCH_CHECK(q.dim() == 4, "flash_prefill: q must be [B, S, Hq, DH]");
  const int B = (int)q.size(0), S = (int)q.size(1), Hq = (int)q.size(2),
            DH = (int)q.size(3);
  const int Hkv = (int)k.size(2);
  TORCH_CHECK(Hq % Hkv == 0, "flash_prefill: Hq % Hkv");
  auto out = at::empty_like(q);
  auto stream = at::hip::getCurrentHIPStream();
  const dim3 grid((S + 255) / 256, Hq, B);

  if (DH == 128) {
    constexpr size_t shmem = KVB * (128 * 2 + 16) + 128 * (KVB * 2 + 16);
    hipLaunchKernelGGL((flash_prefill_kernel<128>), grid, dim3(BLOCK), shmem,
                       stream, (const uint16_t *)q.data_ptr(),
                       (const uint16_t *)k.data_ptr(),
                       (const uint16_t *)v.data_ptr(), (uint16_t *)out.data_ptr(),
                       lsep, B, S, Hq, Hkv, (float)scale);
  } else if (DH == 64) {
    constexpr size_t shmem = KVB * (64 * 2 + 16) + 64 * (KVB * 2 + 16);
    hipLaunchKernelGGL((flash_prefill_kernel<64>), grid, dim3(BLOCK), shmem,
                       stream, (const uint16_t *)q.data_ptr(),
                       (const uint16_t *)k.data_ptr(),
                       (const uint16_t *)v.data_ptr(), (uint16_t *)out.data_ptr(),
                       lsep, B, S, Hq, Hkv, (float)scale);
  } else if (DH == 256) {
    // gemma-7b wide heads: 70.7 KB LDS, fits; closes the last eager
    // prefill fallback (round-1 NOTES item).
    constexpr size_t shmem = KVB * (256 * 2 + 16) + 256 * (KVB * 2 + 16);
    hipLaunchKernelGGL((flash_prefill_kernel<256>), grid, dim3(BLOCK), shmem,
                       stream, (const uint16_t *)q.data_ptr(),
                       (const uint16_t *)k.data_ptr(),
                       (const uint16_t *)v.data_ptr(), (uint16_t *)out.data_ptr(),
                       lsep, B, S, Hq, Hkv, (float)scale);
  } else {
    TORCH_CHECK(false, "flash_prefill: DH must be 64, 128 or 256, got ", DH);
  }
  return out;
}

at::Tensor flash_prefill(at::Tensor q, at::Tensor k, at::Tensor v,
                         double scale) {
  return flash_fwd_impl(q, k, v, scale, nullptr);
}

// Training forward: additionally returns the per-row log-sum-exp
// ([B, Hq, S] f32) consumed by fa_bwd (attention_bwd.hip).
std::vector<at::Tensor> flash_fwd_train(at::Tensor q, at::Tensor k,
                                        at::Tensor v, double scale) {
  const int B = (int)q.size(0), S = (int)q.size(1), Hq = (int)q.size(2);
  auto lse = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  auto out = flash_fwd_impl(q, k, v, scale, lse.data_ptr<float>());
  return {out, lse};
}

// ---------------------------------------------------------------------------
// Layout probe: one v_mfma_f32_32x32x16_bf16 computing C = A @ B with the
// fragment maps this file assumes. The GPU test compares it against torch
// matmul — if the assumed lane->element maps are wrong, THIS fails first
// and pinpoints the bug (instead of the whole flash kernel).
// a: [32, 16] bf16 (A[i][k]);  b: [16, 32] bf16 (B[k][j]) -> c: [32, 32] f32
// ---------------------------------------------------------------------------
namespace {
__global__ void mfma_probe_kernel(const uint16_t *__restrict__ a,
                                  const uint16_t *__restrict__ b,
                                  float *__restrict__ c) {
  const int lane = threadIdx.x & 63;
  const int hi = lane >> 5;
  const int i = lane & 31;
  union { unsigned short u[8]; bf16x8v v; } af, bf;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    af.u[e] = a[i * 16 + hi * 8 + e];         // A[i][k], k = hi*8+e
    bf.u[e] = b[(hi * 8 + e) * 32 + i];       // B[k][j], j = lane&31
  }
  f32x16v acc = (f32x16v)(0.0f);
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af.v, bf.v, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    c[row * 32 + i] = acc[r];                  // C[row][j = lane&31]
  }
}
}  // namespace

at::Tensor mfma_probe_32x32x16(at::Tensor a, at::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16);
  TORCH_CHECK(a.sizes() == at::IntArrayRef({32, 16}) &&
              b.sizes() == at::IntArrayRef({16, 32}), "probe shapes");
  auto c = at::zeros({32, 32}, a.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const uint16_t *)a.contiguous().data_ptr(),
                     (const uint16_t *)b.contiguous().data_ptr(),
                     c.data_ptr<float>());
  return c;
}
