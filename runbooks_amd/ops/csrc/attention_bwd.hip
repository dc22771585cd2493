// Flash-attention TRAINING backward (causal, GQA, bf16) for CDNA4 (gfx950).
//
// FA2-style recompute backward in three dispatches:
//   1. fa_bwd_preprocess: D[b,h,q] = rowsum(dO . O)            (f32)
//   2. fa_bwd_dq:   per 256-query WG, loop KV tiles:
//        S^T = K(scaled Q)^T           mfma(A=K_lds,  B=qf)     C[key][qrow]
//        P^T = exp(S^T - LSE[qrow])
//        dP^T = V dO^T                 mfma(A=V_lds,  B=dof)    C[key][qrow]
//        dS^T = P^T (dP^T - D[qrow]) * scale
//        dQ  += (dS^T)^T K             mfma(A=KT_lds, B=repack(dS^T))
//                                                               C[d][qrow]
//   3. fa_bwd_dkdv: per 256-key WG, loop Q tiles (>= diag):
//        S   = (Q)(scaled K)^T         mfma(A=Q_lds,  B=kf)     C[qrow][key]
//        P   = exp(S - LSE[qrow])
//        dV += P^T dO                  mfma(A=dOT_lds, B=repack(P))
//                                                               C[d][key]
//        dP  = dO V^T                  mfma(A=dO_lds, B=vf)     C[qrow][key]
//        dS  = P (dP - D[qrow]) * scale
//        dK += dS^T Q                  mfma(A=QT_lds, B=repack(dS))
//                                                               C[d][key]
//
// Same wave geometry as the forward (attention_prefill.hip): 8 waves per
// WG, each wave owns 32 rows of its output on lanes (j = lane&31), the
// MFMA C reg dim carries the other axis. The C->B "repack" (pack_bf16 +
// v_permlane32_swap) is the forward's P repack. GQA: dK/dV are computed
// per q-head ([B,S,Hq,DH]); the group-sum to Hkv heads happens in the
// Python wrapper (ops/attention.py) — llama2-7b is MHA so the common path
// has no extra reduction.
//
// mfma_f32_32x32x16_bf16 layouts (cdna_hip_programming.md §3):
//   A[i][k]: i = lane&31, k = (lane>>5)*8 + e
//   B[k][j]: j = lane&31, k = (lane>>5)*8 + e
//   C[i][j]: j = lane&31, i = (r&3) + 8*(r>>2) + 4*(lane>>5)

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;  // 4 waves (128-row WGs: finer grid
                            // halves the causal-triangle imbalance and
                            // doubles WG count vs 8-wave WGs)
constexpr int TILE = 64;    // rows staged to LDS per loop step

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(16))) float f32x16v;

RB_DEV unsigned pack2_bf16(float lo, float hi) {
  union { __bf16 b; unsigned short u; } a, b;
  a.b = (__bf16)lo;
  b.b = (__bf16)hi;
  return ((unsigned)b.u << 16) | a.u;
}

RB_DEV bf16x8v words_to_frag(unsigned w0, unsigned w1, unsigned w2,
                             unsigned w3) {
  union { unsigned u[4]; bf16x8v v; } c;
  c.u[0] = w0; c.u[1] = w1; c.u[2] = w2; c.u[3] = w3;
  return c.v;
}

// C-layout values (16 regs = 32 i-rows) -> two B fragments along k.
// Identical to the forward's P repack; kslot selects k in [0,16)/[16,32).
RB_DEV bf16x8v repack_c_to_b(const float *p, int kslot) {
  const int r0 = kslot * 8;
  unsigned pk0 = pack2_bf16(p[r0 + 0], p[r0 + 1]);
  unsigned pk1 = pack2_bf16(p[r0 + 2], p[r0 + 3]);
  unsigned pk2 = pack2_bf16(p[r0 + 4], p[r0 + 5]);
  unsigned pk3 = pack2_bf16(p[r0 + 6], p[r0 + 7]);
  auto r02 = __builtin_amdgcn_permlane32_swap(pk0, pk2, false, false);
  auto r13 = __builtin_amdgcn_permlane32_swap(pk1, pk3, false, false);
  return words_to_frag(r02[0], r13[0], r02[1], r13[1]);
}

// ---------------------------------------------------------------------------
// 1. D = rowsum(dO . O): dO, O [B, S, H, DH] bf16 -> D [B, H, S] f32
// ---------------------------------------------------------------------------
template <int DH>
__global__ void fa_bwd_preprocess_kernel(const uint16_t *__restrict__ dop,
                                         const uint16_t *__restrict__ op,
                                         float *__restrict__ dp,
                                         int B, int S, int H) {
  constexpr int LPR = DH / 32;          // 16B vectors per lane chunk
  const int row = blockIdx.x * 64 + (threadIdx.x >> 2);  // global (b*S+s)*H+h
  const int c = threadIdx.x & 3;
  const int64_t total = (int64_t)B * S * H;
  if (row >= total) return;
  const uint16_t *o = op + (int64_t)row * DH + c * (DH / 4);
  const uint16_t *g = dop + (int64_t)row * DH + c * (DH / 4);
  float acc = 0.0f;
#pragma unroll
  for (int v = 0; v < LPR; ++v) {
    float fo[8], fg[8];
    rb::VIO<uint16_t>::load(o + v * 8, fo);
    rb::VIO<uint16_t>::load(g + v * 8, fg);
#pragma unroll
    for (int e = 0; e < 8; ++e) acc += fo[e] * fg[e];
  }
#pragma unroll
  for (int off = 1; off < 4; off <<= 1) acc += __shfl_xor(acc, off, 64);
  if (c == 0) {
    const int h = row % H;
    const int s = (row / H) % S;
    const int b = row / (H * (int64_t)S);
    dp[((int64_t)b * H + h) * S + s] = acc;
  }
}

// ---------------------------------------------------------------------------
// 2. dQ kernel — mirrors the forward loop over KV tiles.
// q, k, v, dout: [B, S, H*, DH] bf16; lse, dvec: [B, Hq, S] f32.
// dq: [B, S, Hq, DH] bf16.
// ---------------------------------------------------------------------------
template <int DH>
__global__ __launch_bounds__(BLOCK, 2) void fa_bwd_dq_kernel(
    const uint16_t *__restrict__ qp, const uint16_t *__restrict__ kp,
    const uint16_t *__restrict__ vp, const uint16_t *__restrict__ dop,
    const float *__restrict__ lsep, const float *__restrict__ dvecp,
    uint16_t *__restrict__ dqp, int B, int S, int Hq, int Hkv, float scale) {
  constexpr int KSTEPS = DH / 16;
  constexpr int DTILES = DH / 32;
  constexpr int ROW_STRIDE = DH * 2 + 16;   // row-major tile row bytes
  constexpr int TR_STRIDE = TILE * 2 + 16;  // transposed tile row bytes

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char *k_img = smem;                           // [TILE][ROW_STRIDE]
  char *v_img = k_img + TILE * ROW_STRIDE;      // [TILE][ROW_STRIDE]
  char *kt_img = v_img + TILE * ROW_STRIDE;     // [DH][TR_STRIDE]

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;
  const int col = lane & 31;

  const int q0 = blockIdx.x * 128;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int h_kv = h / (Hq / Hkv);
  const int qrow = q0 + wid * 32 + col;
  const bool q_valid = qrow < S;

  // per-lane fragments for this lane's query row
  bf16x8v qf[KSTEPS];   // scaled Q (B operand of S^T)
  bf16x8v dof[KSTEPS];  // dO (B operand of dP^T)
  float lse = q_valid ? lsep[((int64_t)b * Hq + h) * S + qrow] : INFINITY;
  float dvec = q_valid ? dvecp[((int64_t)b * Hq + h) * S + qrow] : 0.0f;
  {
    const int64_t base = ((int64_t)(b * S + (q_valid ? qrow : 0)) * Hq + h) * DH;
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      float fq[8], fd[8];
      rb::VIO<uint16_t>::load(qp + base + ks * 16 + hi * 8, fq);
      rb::VIO<uint16_t>::load(dop + base + ks * 16 + hi * 8, fd);
      union { unsigned short u[8]; bf16x8v v; } cq, cd;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        cq.u[e] = rb::f32_to_bf16(q_valid ? fq[e] * scale : 0.0f);
        cd.u[e] = rb::f32_to_bf16(q_valid ? fd[e] : 0.0f);
      }
      qf[ks] = cq.v;
      dof[ks] = cd.v;
    }
  }

  f32x16v acc_dq[DTILES];
#pragma unroll
  for (int dt = 0; dt < DTILES; ++dt) acc_dq[dt] = (f32x16v)(0.0f);

  const int kv_end = min(S, q0 + 128);
  const int n_tiles = (kv_end + TILE - 1) / TILE;
  const int wave_kmax = q0 + wid * 32 + 31;

  for (int kt = 0; kt < n_tiles; ++kt) {
    const int kbase = kt * TILE;

    // ---- stage K and V row-major ----------------------------------------
    {
      constexpr int total = TILE * DH;
      const int per_pass = BLOCK * 8;
#pragma unroll
      for (int p = 0; p < total / per_pass; ++p) {
        const int idx = p * per_pass + tid * 8;
        const int key = idx / DH;
        const int d0 = idx % DH;
        const int gk = kbase + key;
        rb::bf16x8 kv8, vv8;
        if (gk < S) {
          const int64_t base = ((int64_t)(b * S + gk) * Hkv + h_kv) * DH + d0;
          kv8 = *reinterpret_cast<const rb::bf16x8 *>(kp + base);
          vv8 = *reinterpret_cast<const rb::bf16x8 *>(vp + base);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) { kv8.v[e] = 0; vv8.v[e] = 0; }
        }
        *reinterpret_cast<rb::bf16x8 *>(k_img + key * ROW_STRIDE + d0 * 2) = kv8;
        *reinterpret_cast<rb::bf16x8 *>(v_img + key * ROW_STRIDE + d0 * 2) = vv8;
      }
    }
    // ---- stage K transposed ([d][key]) -----------------------------------
    // from the LDS row image (same fix as dkdv: the global column-
    // strided scalar re-read was latency-bound)
    __syncthreads();
    {
      constexpr int GRPS = BLOCK / DH;
      constexpr int KPG = TILE / GRPS;
      const int d = tid % DH;
      const int kg0 = (tid / DH) * KPG;
      uint16_t tmp[KPG];
#pragma unroll
      for (int e = 0; e < KPG; ++e) {
        tmp[e] = *reinterpret_cast<const uint16_t *>(
            k_img + (kg0 + e) * ROW_STRIDE + d * 2);
      }
#pragma unroll
      for (int c8 = 0; c8 < KPG / 8; ++c8)
        *reinterpret_cast<rb::bf16x8 *>(
            kt_img + d * TR_STRIDE + (kg0 + c8 * 8) * 2) =
            *reinterpret_cast<rb::bf16x8 *>(&tmp[c8 * 8]);
    }
    __syncthreads();

    if (kbase <= wave_kmax) {
#pragma unroll
      for (int it = 0; it < 2; ++it) {
        // ---- S^T[key][qrow] and dP^T[key][qrow] --------------------------
        f32x16v s = (f32x16v)(0.0f);
        f32x16v dpt = (f32x16v)(0.0f);
#pragma unroll
        for (int ks = 0; ks < KSTEPS; ++ks) {
          const bf16x8v kfr = *reinterpret_cast<const bf16x8v *>(
              k_img + (it * 32 + col) * ROW_STRIDE + (ks * 16 + hi * 8) * 2);
          const bf16x8v vfr = *reinterpret_cast<const bf16x8v *>(
              v_img + (it * 32 + col) * ROW_STRIDE + (ks * 16 + hi * 8) * 2);
          s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfr, qf[ks], s, 0, 0, 0);
          dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfr, dof[ks], dpt,
                                                        0, 0, 0);
        }

        // ---- dS^T = P^T (dP^T - D) * scale -------------------------------
        float ds[16];
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int key = kbase + it * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          const float pt = (key <= qrow) ? __expf(s[r] - lse) : 0.0f;
          ds[r] = pt * (dpt[r] - dvec) * scale;
        }

        // ---- dQ += (dS^T)^T K -------------------------------------------
#pragma unroll
        for (int kslot = 0; kslot < 2; ++kslot) {
          const bf16x8v dsf = repack_c_to_b(ds, kslot);
          const int koff = (it * 32 + kslot * 16 + hi * 8) * 2;
#pragma unroll
          for (int dt = 0; dt < DTILES; ++dt) {
            const bf16x8v ktf = *reinterpret_cast<const bf16x8v *>(
                kt_img + (dt * 32 + col) * TR_STRIDE + koff);
            acc_dq[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                ktf, dsf, acc_dq[dt], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: dQ[b, qrow, h, :] ----------------------------------------
  if (q_valid) {
    uint16_t *row = dqp + ((int64_t)(b * S + qrow) * Hq + h) * DH;
#pragma unroll
    for (int dt = 0; dt < DTILES; ++dt) {
#pragma unroll
      for (int rq = 0; rq < 4; ++rq) {
        const int d0 = dt * 32 + 8 * rq + 4 * hi;
        uint16_t w[4];
#pragma unroll
        for (int e = 0; e < 4; ++e)
          w[e] = rb::f32_to_bf16(acc_dq[dt][rq * 4 + e]);
        *reinterpret_cast<uint2 *>(row + d0) =
            *reinterpret_cast<const uint2 *>(w);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// 3. dK/dV kernel — wave owns 32 keys, loops over Q tiles at/after the
// diagonal. Outputs per-q-head dk, dv [B, S, Hq, DH] bf16.
// ---------------------------------------------------------------------------
template <int DH>
__global__ __launch_bounds__(BLOCK, 2) void fa_bwd_dkdv_kernel(
    const uint16_t *__restrict__ qp, const uint16_t *__restrict__ kp,
    const uint16_t *__restrict__ vp, const uint16_t *__restrict__ dop,
    const float *__restrict__ lsep, const float *__restrict__ dvecp,
    uint16_t *__restrict__ dkp, uint16_t *__restrict__ dvp,
    int B, int S, int Hq, int Hkv, float scale) {
  constexpr int KSTEPS = DH / 16;
  constexpr int DTILES = DH / 32;
  constexpr int ROW_STRIDE = DH * 2 + 16;
  constexpr int TR_STRIDE = TILE * 2 + 16;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char *q_img = smem;                            // [TILE][ROW_STRIDE]
  char *do_img = q_img + TILE * ROW_STRIDE;      // [TILE][ROW_STRIDE]
  char *qt_img = do_img + TILE * ROW_STRIDE;     // [DH][TR_STRIDE]
  char *dot_img = qt_img + DH * TR_STRIDE;       // [DH][TR_STRIDE]
  float *lse_s = reinterpret_cast<float *>(dot_img + DH * TR_STRIDE);  // [TILE]
  float *dvec_s = lse_s + TILE;                                        // [TILE]

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;
  const int col = lane & 31;

  const int k0 = blockIdx.x * 128;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int h_kv = h / (Hq / Hkv);
  const int key = k0 + wid * 32 + col;       // this lane's key row
  const bool k_valid = key < S;

  // per-lane fragments for this lane's key row
  bf16x8v kf[KSTEPS];   // scaled K (B operand of S)
  bf16x8v vf[KSTEPS];   // V (B operand of dP)
  {
    const int64_t base =
        ((int64_t)(b * S + (k_valid ? key : 0)) * Hkv + h_kv) * DH;
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
      float fk[8], fv[8];
      rb::VIO<uint16_t>::load(kp + base + ks * 16 + hi * 8, fk);
      rb::VIO<uint16_t>::load(vp + base + ks * 16 + hi * 8, fv);
      union { unsigned short u[8]; bf16x8v v; } ck, cv;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        ck.u[e] = rb::f32_to_bf16(k_valid ? fk[e] * scale : 0.0f);
        cv.u[e] = rb::f32_to_bf16(k_valid ? fv[e] : 0.0f);
      }
      kf[ks] = ck.v;
      vf[ks] = cv.v;
    }
  }

  f32x16v acc_dk[DTILES], acc_dv[DTILES];
#pragma unroll
  for (int dt = 0; dt < DTILES; ++dt) {
    acc_dk[dt] = (f32x16v)(0.0f);
    acc_dv[dt] = (f32x16v)(0.0f);
  }

  const int wave_kmin = k0 + wid * 32;       // first key this wave owns
  const int n_tiles = (S - k0 + TILE - 1) / TILE;

  for (int qt = 0; qt < n_tiles; ++qt) {
    const int qbase = k0 + qt * TILE;

    // ---- stage Q and dO row-major ----------------------------------------
    {
      constexpr int total = TILE * DH;
      const int per_pass = BLOCK * 8;
#pragma unroll
      for (int p = 0; p < total / per_pass; ++p) {
        const int idx = p * per_pass + tid * 8;
        const int qr = idx / DH;
        const int d0 = idx % DH;
        const int gq = qbase + qr;
        rb::bf16x8 q8, d8;
        if (gq < S) {
          const int64_t base = ((int64_t)(b * S + gq) * Hq + h) * DH + d0;
          q8 = *reinterpret_cast<const rb::bf16x8 *>(qp + base);
          d8 = *reinterpret_cast<const rb::bf16x8 *>(dop + base);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) { q8.v[e] = 0; d8.v[e] = 0; }
        }
        *reinterpret_cast<rb::bf16x8 *>(q_img + qr * ROW_STRIDE + d0 * 2) = q8;
        *reinterpret_cast<rb::bf16x8 *>(do_img + qr * ROW_STRIDE + d0 * 2) = d8;
      }
    }
    // ---- stage Q^T and dO^T ([d][qrow]) ----------------------------------
    // Sourced from the just-staged LDS row images (50-cycle LDS reads)
    // instead of re-reading global column-strided with scalar loads
    // (~900-cycle HBM/L2 round trips per element — measured r9: dkdv at
    // 235 us was the largest single train kernel).
    __syncthreads();
    {
      constexpr int GRPS = BLOCK / DH;
      constexpr int KPG = TILE / GRPS;
      const int d = tid % DH;
      const int qg0 = (tid / DH) * KPG;
      uint16_t tq[KPG], td[KPG];
#pragma unroll
      for (int e = 0; e < KPG; ++e) {
        const int qr = qg0 + e;
        tq[e] = *reinterpret_cast<const uint16_t *>(
            q_img + qr * ROW_STRIDE + d * 2);
        td[e] = *reinterpret_cast<const uint16_t *>(
            do_img + qr * ROW_STRIDE + d * 2);
      }
#pragma unroll
      for (int c8 = 0; c8 < KPG / 8; ++c8) {
        *reinterpret_cast<rb::bf16x8 *>(
            qt_img + d * TR_STRIDE + (qg0 + c8 * 8) * 2) =
            *reinterpret_cast<rb::bf16x8 *>(&tq[c8 * 8]);
        *reinterpret_cast<rb::bf16x8 *>(
            dot_img + d * TR_STRIDE + (qg0 + c8 * 8) * 2) =
            *reinterpret_cast<rb::bf16x8 *>(&td[c8 * 8]);
      }
    }
    // ---- stage per-qrow LSE and D ----------------------------------------
    if (tid < TILE) {
      const int gq = qbase + tid;
      if (gq < S) {
        lse_s[tid] = lsep[((int64_t)b * Hq + h) * S + gq];
        dvec_s[tid] = dvecp[((int64_t)b * Hq + h) * S + gq];
      } else {
        lse_s[tid] = INFINITY;
        dvec_s[tid] = 0.0f;
      }
    }
    __syncthreads();

#pragma unroll
    for (int it = 0; it < 2; ++it) {
      // this 32-qrow subtile reaches the wave's keys?
      if (qbase + it * 32 + 31 >= wave_kmin) {
        // ---- S[qrow][key] and dP[qrow][key] ------------------------------
        f32x16v s = (f32x16v)(0.0f);
        f32x16v dp = (f32x16v)(0.0f);
#pragma unroll
        for (int ks = 0; ks < KSTEPS; ++ks) {
          const bf16x8v qfr = *reinterpret_cast<const bf16x8v *>(
              q_img + (it * 32 + col) * ROW_STRIDE + (ks * 16 + hi * 8) * 2);
          const bf16x8v dfr = *reinterpret_cast<const bf16x8v *>(
              do_img + (it * 32 + col) * ROW_STRIDE + (ks * 16 + hi * 8) * 2);
          s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qfr, kf[ks], s, 0, 0, 0);
          dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dfr, vf[ks], dp,
                                                       0, 0, 0);
        }

        // ---- P and dS (both in C layout over qrow regs) ------------------
        float pv[16], ds[16];
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qr = it * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          const int gq = qbase + qr;
          const float p = (gq >= key) ? __expf(s[r] - lse_s[qr]) : 0.0f;
          pv[r] = p;
          ds[r] = p * (dp[r] - dvec_s[qr]) * scale;
        }

        // ---- dV += P^T dO; dK += dS^T Q ----------------------------------
#pragma unroll
        for (int kslot = 0; kslot < 2; ++kslot) {
          const bf16x8v pf = repack_c_to_b(pv, kslot);
          const bf16x8v dsf = repack_c_to_b(ds, kslot);
          const int qoff = (it * 32 + kslot * 16 + hi * 8) * 2;
#pragma unroll
          for (int dt = 0; dt < DTILES; ++dt) {
            const bf16x8v dotf = *reinterpret_cast<const bf16x8v *>(
                dot_img + (dt * 32 + col) * TR_STRIDE + qoff);
            const bf16x8v qtf = *reinterpret_cast<const bf16x8v *>(
                qt_img + (dt * 32 + col) * TR_STRIDE + qoff);
            acc_dv[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                dotf, pf, acc_dv[dt], 0, 0, 0);
            acc_dk[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                qtf, dsf, acc_dk[dt], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: dK, dV -> [b, key, h, :] ---------------------------------
  if (k_valid) {
    const int64_t base = ((int64_t)(b * S + key) * Hq + h) * DH;
#pragma unroll
    for (int dt = 0; dt < DTILES; ++dt) {
#pragma unroll
      for (int rq = 0; rq < 4; ++rq) {
        const int d0 = dt * 32 + 8 * rq + 4 * hi;
        uint16_t wk[4], wv[4];
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          wk[e] = rb::f32_to_bf16(acc_dk[dt][rq * 4 + e]);
          wv[e] = rb::f32_to_bf16(acc_dv[dt][rq * 4 + e]);
        }
        *reinterpret_cast<uint2 *>(dkp + base + d0) =
            *reinterpret_cast<const uint2 *>(wk);
        *reinterpret_cast<uint2 *>(dvp + base + d0) =
            *reinterpret_cast<const uint2 *>(wv);
      }
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

std::vector<at::Tensor> fa_bwd(at::Tensor dout, at::Tensor q, at::Tensor k,
                               at::Tensor v, at::Tensor out, at::Tensor lse,
                               double scale) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous() &&
              v.is_contiguous() && dout.is_contiguous() && out.is_contiguous(),
              "fa_bwd: contiguous GPU tensors");
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "fa_bwd: bf16 only");
  const int B = (int)q.size(0), S = (int)q.size(1), Hq = (int)q.size(2),
            DH = (int)q.size(3);
  const int Hkv = (int)k.size(2);
  TORCH_CHECK(DH == 64 || DH == 128, "fa_bwd: DH must be 64 or 128");
  TORCH_CHECK(lse.scalar_type() == at::kFloat && lse.is_contiguous());

  auto dq = at::empty_like(q);
  // per-q-head dk/dv; group-summed by the caller when Hkv < Hq
  auto dk = at::empty({B, S, Hq, DH}, q.options());
  auto dv = at::empty({B, S, Hq, DH}, q.options());
  auto dvec = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));

  auto stream = at::hip::getCurrentHIPStream();
  {
    const int64_t rows = (int64_t)B * S * Hq;
    const dim3 grid((unsigned)((rows + 63) / 64));
    if (DH == 128)
      hipLaunchKernelGGL((fa_bwd_preprocess_kernel<128>), grid, dim3(256), 0,
                         stream, (const uint16_t *)dout.data_ptr(),
                         (const uint16_t *)out.data_ptr(),
                         dvec.data_ptr<float>(), B, S, Hq);
    else
      hipLaunchKernelGGL((fa_bwd_preprocess_kernel<64>), grid, dim3(256), 0,
                         stream, (const uint16_t *)dout.data_ptr(),
                         (const uint16_t *)out.data_ptr(),
                         dvec.data_ptr<float>(), B, S, Hq);
  }

  const dim3 grid((S + 127) / 128, Hq, B);
#define RB_LAUNCH_BWD(DHV)                                                    \
  do {                                                                        \
    constexpr int ROW_STRIDE = DHV * 2 + 16;                                  \
    constexpr int TR_STRIDE = TILE * 2 + 16;                                  \
    constexpr size_t sh_dq =                                                  \
        2 * TILE * ROW_STRIDE + DHV * TR_STRIDE;                              \
    constexpr size_t sh_dkdv = 2 * TILE * ROW_STRIDE +                        \
        2 * DHV * TR_STRIDE + 2 * TILE * sizeof(float);                       \
    hipLaunchKernelGGL((fa_bwd_dq_kernel<DHV>), grid, dim3(BLOCK), sh_dq,     \
                       stream, (const uint16_t *)q.data_ptr(),                \
                       (const uint16_t *)k.data_ptr(),                        \
                       (const uint16_t *)v.data_ptr(),                        \
                       (const uint16_t *)dout.data_ptr(),                     \
                       lse.data_ptr<float>(), dvec.data_ptr<float>(),         \
                       (uint16_t *)dq.data_ptr(), B, S, Hq, Hkv,              \
                       (float)scale);                                         \
    hipLaunchKernelGGL((fa_bwd_dkdv_kernel<DHV>), grid, dim3(BLOCK), sh_dkdv, \
                       stream, (const uint16_t *)q.data_ptr(),                \
                       (const uint16_t *)k.data_ptr(),                        \
                       (const uint16_t *)v.data_ptr(),                        \
                       (const uint16_t *)dout.data_ptr(),                     \
                       lse.data_ptr<float>(), dvec.data_ptr<float>(),         \
                       (uint16_t *)dk.data_ptr(), (uint16_t *)dv.data_ptr(),  \
                       B, S, Hq, Hkv, (float)scale);                          \
  } while (0)

  if (DH == 128) RB_LAUNCH_BWD(128);
  else RB_LAUNCH_BWD(64);
#undef RB_LAUNCH_BWD
  return {dq, dk, dv};
}
