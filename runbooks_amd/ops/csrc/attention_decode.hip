// Paged decode attention (one query token per sequence) for CDNA4 (gfx950).
//
// The serving hot loop: memory-bound on the KV read (288 GB HBM3E at
// ~6.3 TB/s achievable). Design, MI355X-first:
//  * wave64 split into four 16-lane groups; each group owns ONE kv token
//    per step at 8 (Dh=128) or 4 (Dh=64) elems/lane, so every wave streams
//    4 contiguous KV rows per iteration at 16/8 B per lane.
//  * GQA/MQA-native: a workgroup handles one (seq, kv_head); its G query
//    heads share each K/V read (falcon-40b MQA G=16, llama2-70b G=8,
//    llama2-7b MHA G=1 — SURVEY.md §2b "server image").
//  * flash-style online softmax entirely in registers; group partials
//    merged via shfl_xor(16/32), wave partials via LDS.
//  * kv-split ("flash-decoding") for long sequences / small batch: grid z
//    = nsplit partitions, fp32 partials (m, l, acc) merged by a tiny
//    second kernel — sized so batch*Hkv*nsplit >> 256 CUs.
//
// q:       [B, Hq, Dh] bf16     out: [B, Hq, Dh] bf16
// caches:  [num_blocks, Hkv, BS, Dh] bf16 (see kvcache.hip)
// block_tables: [B, max_blocks] int32;  seq_lens: [B] int32
// partial: [B, Hq, nsplit, Dh + 2] fp32 workspace when nsplit > 1

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;    // 4 waves
constexpr int NW = 4;

typedef __attribute__((ext_vector_type(8))) __bf16 rb_bf16x8v;
typedef __attribute__((ext_vector_type(2))) __bf16 rb_bf16x2v;
typedef __attribute__((ext_vector_type(16))) float rb_f32x16v;
typedef __attribute__((ext_vector_type(4))) float rb_f32x4v;

RB_DEV unsigned rb_pack_bf16(float lo, float hi) {
  union { __bf16 b; unsigned short u; } a, b;
  a.b = (__bf16)lo;
  b.b = (__bf16)hi;
  return ((unsigned)b.u << 16) | a.u;
}

// one v_dot2_f32_bf16: c += a_pair . b_pair (packed bf16 pairs)
RB_DEV float rb_dot2(unsigned a, unsigned b, float c) {
  union { unsigned u; rb_bf16x2v v; } x, y;
  x.u = a;
  y.u = b;
  return __builtin_amdgcn_fdot2_f32_bf16(x.v, y.v, c, false);
}

// FP8: caches are e4m3 rows of fp8_row_bytes(DH) = DH bytes + f32
// scale (+pad): decode attention is KV-bandwidth bound at long context,
// so halving cache bytes halves the dominant stream AND doubles KV
// capacity within 288 GB (BASELINE config #5). Dequant rides
// v_cvt_pk_f32_fp8 in the inner loop.
template <int DH, int G, bool SPLIT, bool FP8 = false>
__global__ __launch_bounds__(BLOCK) void paged_decode_kernel(
    const uint16_t *__restrict__ q, const uint16_t *__restrict__ k_cache,
    const uint16_t *__restrict__ v_cache, const int32_t *__restrict__ block_tables,
    const int32_t *__restrict__ seq_lens,
    const int32_t *__restrict__ seq_starts, uint16_t *__restrict__ out,
    uint16_t *__restrict__ out_swz,
    float *__restrict__ partial, int hkv, int bs, int max_blocks, int nsplit,
    float scale) {
  // Lane-group geometry: GL lanes cover one token's Dh. bf16 reads 2 B
  // per element, so 16 lanes x VE=DH/16 elems = 16 B/lane; fp8 reads
  // 1 B per element, so 8-lane groups with VE=DH/8 keep the loads at
  // the full 16 B width (8 B accesses run 0.54-0.70x on gfx950 — the
  // first fp8 cut used 16-lane groups and showed ZERO long-context win).
  // (G > 4 keeps 16-lane groups even for fp8: qreg[G][VE]+acc[G][VE]
  // at VE=16 would spill past 256 VGPRs)
  constexpr int GL = (FP8 && G <= 4) ? 8 : 16;
  constexpr int VE = DH / GL;              // elems per lane
  constexpr int TPW = 64 / GL;             // tokens per wave pass
  // Head-split mode (G >= NW): each wave owns GW = G/NW query heads
  // over ALL tokens — the waves re-read the same K/V rows (L1/L2-
  // served; HBM bytes unchanged) — instead of all G heads over a
  // quarter of the tokens. Cuts per-wave state (qreg+acc = 2*G*VE
  // floats) by NW, unlocking occupancy 2 -> 4-5 on the GQA/MQA
  // kernels, shortens the serial dot->reduce->exp chain per
  // iteration, and removes the cross-wave LDS merge entirely.
  constexpr bool HS = (G >= NW);
  constexpr int GW = HS ? G / NW : G;
  const int b = blockIdx.x;
  const int h_kv = blockIdx.y;
  const int split = SPLIT ? blockIdx.z : 0;
  const int hq0 = h_kv * G;
  const int Hq = hkv * G;

  const int seq_len = seq_lens[b];
  // Strict sliding windows: seq_starts[b] (0 when null) is the first
  // VIRTUAL position this query attends to — mistral attends to exactly
  // W trailing tokens, not the engine's retained block-aligned W..W+15.
  // Index math pre-validated on CPU: tests/test_decode_sim_cpu.py
  // (simulate_decode seq_starts).
  const int start = (seq_starts != nullptr) ? seq_starts[b] : 0;
  int t_begin = start, t_end = seq_len;
  if (SPLIT) {
    const int len = seq_len - start;
    const int chunk = (len + nsplit - 1) / nsplit;
    t_begin = start + split * chunk;
    t_end = min(seq_len, t_begin + chunk);
  }

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int grp = lane / GL;               // lane group: token sub-index
  const int gl = lane % GL;                // lane within group
  const int d0 = gl * VE;                  // this lane's Dh slice

  // Q for this wave's GW heads, pre-scaled (softmax scale folded in).
  // bf16 caches keep Q as PACKED bf16 pairs too: the K dot then runs on
  // v_dot2_f32_bf16 straight from the raw K words — no per-element
  // convert+FMA chain (the kernel is VALU-saturated at occupancy 8;
  // the same q*scale->bf16 rounding the MFMA kernel already makes).
  const int hq_base = hq0 + (HS ? wid * GW : 0);
  float qreg[GW][VE];
  unsigned qpk[GW][VE / 2];
#pragma unroll
  for (int g = 0; g < GW; ++g) {
    const uint16_t *qp = q + ((int64_t)b * Hq + hq_base + g) * DH + d0;
#pragma unroll
    for (int e = 0; e < VE; ++e) qreg[g][e] = 0.f;
    {
      float tmp[VE];
#pragma unroll
      for (int e8 = 0; e8 < VE; e8 += 8) {
        if (VE - e8 >= 8) rb::VIO<uint16_t>::load(qp + e8, tmp + e8);
        else {
#pragma unroll
          for (int e = 0; e < VE % 8; ++e)
            tmp[e8 + e] = rb::bf16_to_f32(qp[e8 + e]);
        }
      }
      if (FP8) {
#pragma unroll
        for (int e = 0; e < VE; ++e) qreg[g][e] = tmp[e] * scale;
      } else {
#pragma unroll
        for (int e2 = 0; e2 < VE / 2; ++e2)
          qpk[g][e2] = rb_pack_bf16(tmp[2 * e2] * scale,
                                    tmp[2 * e2 + 1] * scale);
      }
    }
  }

  float m[GW], l[GW], acc[GW][VE];
#pragma unroll
  for (int g = 0; g < GW; ++g) {
    m[g] = -INFINITY; l[g] = 0.f;
#pragma unroll
    for (int e = 0; e < VE; ++e) acc[g][e] = 0.f;
  }

  const int32_t *bt = block_tables + (int64_t)b * max_blocks;

  // Stage this WG's slice of the block table through LDS: all four
  // waves walk the same row, and a per-iteration global bt[] read
  // sits on the K/V address chain (its vmcnt gates the row loads).
  // 512 entries cover 8K tokens at bs=16; longer ranges fall back to
  // global reads past the staged prefix.
  constexpr int BT_LDS = 512;
  __shared__ int32_t bt_lds[BT_LDS];
  const int blk0 = t_begin / bs;
  if (t_begin < t_end) {
    const int nbt = min((t_end - 1) / bs - blk0 + 1, BT_LDS);
    for (int i = (int)threadIdx.x; i < nbt; i += BLOCK)
      bt_lds[i] = bt[blk0 + i];
  }
  __syncthreads();

  // Software-pipelined KV stream: iteration i+1's K/V rows are issued
  // as RAW bytes while i is computed. (Converting at load time puts an
  // s_waitcnt vmcnt(0) between every row and its dot product — the
  // unpipelined loop measured ~3.9 TB/s effective in-situ vs ~6 TB/s
  // streaming.) Tail iterations clamp to the last row; the duplicate
  // rows are masked out of the softmax with s = -inf, so the trip
  // count is wave-uniform and the prefetch is branch-free inside.
  constexpr int KBPL = FP8 ? VE : VE * 2;  // K/V bytes per lane
  constexpr int RW = KBPL / 4;             // u32 words of that
  // Ring depth: the G>=4 instantiations are VGPR-capped at 2 waves/SIMD
  // (qreg+acc alone are 2*G*VE floats), so two waves must hide the full
  // HBM latency — a 4-deep ring keeps 3 iterations of K/V in flight for
  // only +4*RW VGPRs. G<=2 runs 5-8 waves/SIMD where depth 2 suffices
  // (deeper costs occupancy: 62->78 VGPRs drops the MHA kernel 8->6).
  constexpr int DEPTH = (G >= 4) ? 4 : 2;
  const int stride = HS ? TPW : NW * TPW;
  const int tw0 = t_begin + (HS ? 0 : wid * TPW);  // wave's first token
  const int t_base = tw0 + grp;            // this lane group's first
  const int niter = (t_end > tw0) ? (t_end - tw0 + stride - 1) / stride : 0;

  uint32_t kraw[DEPTH][RW], vraw[DEPTH][RW];
  float kscale[DEPTH], vscale[DEPTH];

  auto fetch = [&](int i, int slot) {
    const int tt = min(t_base + i * stride, t_end - 1);
    const int bi = tt / bs - blk0;
    const int blk = (bi < BT_LDS) ? bt_lds[bi] : bt[tt / bs];
    if (FP8) {
      constexpr int RB8 = DH + 16;
      const uint8_t *k8 = reinterpret_cast<const uint8_t *>(k_cache);
      const uint8_t *v8 = reinterpret_cast<const uint8_t *>(v_cache);
      const int64_t row = (((int64_t)blk * hkv + h_kv) * bs + (tt % bs)) * RB8;
      rb::ld_words<RW>(kraw[slot], k8 + row + d0);
      rb::ld_words<RW>(vraw[slot], v8 + row + d0);
      kscale[slot] = *reinterpret_cast<const float *>(k8 + row + DH);
      vscale[slot] = *reinterpret_cast<const float *>(v8 + row + DH);
    } else {
      const int64_t base =
          (((int64_t)blk * hkv + h_kv) * bs + (tt % bs)) * DH + d0;
      rb::ld_words<RW>(kraw[slot],
                       reinterpret_cast<const uint8_t *>(k_cache + base));
      rb::ld_words<RW>(vraw[slot],
                       reinterpret_cast<const uint8_t *>(v_cache + base));
    }
  };

  // cur is a compile-time constant at every call site below, so the
  // ring buffer stays in registers (no dynamic indexing).
  auto step = [&](int i, int cur) {
    if (i + DEPTH - 1 < niter) fetch(i + DEPTH - 1, (cur + DEPTH - 1) % DEPTH);
    const bool valid = t_base + i * stride < t_end;

    float kf[VE], vf[VE];
    if (FP8) {
#pragma unroll
      for (int w = 0; w < RW; ++w) {
        rb::fp8w_to_f32(kraw[cur][w], kf + w * 4);
        rb::fp8w_to_f32(vraw[cur][w], vf + w * 4);
      }
      const float ks = kscale[cur], vs = vscale[cur];
#pragma unroll
      for (int e = 0; e < VE; ++e) { kf[e] *= ks; vf[e] *= vs; }
    } else {
      // only V needs f32 elements (the p*v accumulate); the K dot runs
      // packed below, straight from the raw words
      const uint16_t *vr = reinterpret_cast<const uint16_t *>(vraw[cur]);
#pragma unroll
      for (int e = 0; e < VE; ++e) vf[e] = rb::bf16_to_f32(vr[e]);
    }

#pragma unroll
    for (int g = 0; g < GW; ++g) {
      float s = 0.f;
      if (FP8) {
#pragma unroll
        for (int e = 0; e < VE; ++e) s += qreg[g][e] * kf[e];
      } else {
#pragma unroll
        for (int e2 = 0; e2 < VE / 2; ++e2)
          s = rb_dot2(kraw[cur][e2], qpk[g][e2], s);
      }
      // reduce across the lane group
#pragma unroll
      for (int off = GL / 2; off > 0; off >>= 1)
        s += __shfl_xor(s, off, 64);
      if (!valid) s = -INFINITY;
      const float mn = fmaxf(m[g], s);
      if (mn != -INFINITY) {
        const float alpha = __expf(m[g] - mn);
        const float p = __expf(s - mn);
        l[g] = l[g] * alpha + p;
#pragma unroll
        for (int e = 0; e < VE; ++e) acc[g][e] = acc[g][e] * alpha + p * vf[e];
        m[g] = mn;
      }
    }
  };

#pragma unroll
  for (int d = 0; d < DEPTH - 1; ++d)
    if (d < niter) fetch(d, d);
  for (int i = 0; i < niter; i += DEPTH) {
    step(i, 0);
    if (i + 1 < niter) step(i + 1, 1 % DEPTH);
    if (DEPTH > 2 && i + 2 < niter) step(i + 2, 2 % DEPTH);
    if (DEPTH > 2 && i + 3 < niter) step(i + 3, 3 % DEPTH);
  }

  // Merge the TPW groups inside each wave: lanes l^GL ... l^32 hold the
  // same d0 slice, so shfl_xor merges matching elements.
#pragma unroll
  for (int off = GL; off <= 32; off <<= 1) {
#pragma unroll
    for (int g = 0; g < GW; ++g) {
      const float mo = __shfl_xor(m[g], off, 64);
      const float lo = __shfl_xor(l[g], off, 64);
      const float mn = fmaxf(m[g], mo);
      if (mn == -INFINITY) continue;
      const float a1 = __expf(m[g] - mn);
      const float a2 = __expf(mo - mn);
#pragma unroll
      for (int e = 0; e < VE; ++e) {
        const float ao = __shfl_xor(acc[g][e], off, 64);
        acc[g][e] = acc[g][e] * a1 + ao * a2;
      }
      l[g] = l[g] * a1 + lo * a2;
      m[g] = mn;
    }
  }

  if constexpr (HS) {
    // Head-split: each wave owns its GW heads outright — write out
    // (or the split partial) straight from registers, no LDS merge.
    if (grp == 0) {
#pragma unroll
      for (int g = 0; g < GW; ++g) {
        const int h = hq_base + g;
        if (SPLIT) {
          float *pp =
              partial + (((int64_t)b * Hq + h) * nsplit + split) * (DH + 2);
#pragma unroll
          for (int e = 0; e < VE; ++e) pp[d0 + e] = acc[g][e];
          if (gl == 0) { pp[DH] = m[g]; pp[DH + 1] = l[g]; }
        } else {
          const float inv_l = (l[g] > 0.f) ? 1.0f / l[g] : 0.f;
          uint16_t *op = out + ((int64_t)b * Hq + h) * DH + d0;
#pragma unroll
          for (int e = 0; e < VE; ++e)
            op[e] = rb::f32_to_bf16(acc[g][e] * inv_l);
          if (VE >= 8 && out_swz != nullptr) {
#pragma unroll
            for (int e8 = 0; e8 < VE; e8 += 8) {
              const int kk = h * DH + d0 + e8;
              uint16_t *sp = out_swz + (kk >> 4) * 512 +
                             ((kk >> 3) & 1) * 256 + b * 8;
#pragma unroll
              for (int e = 0; e < 8 && e < VE; ++e)
                sp[e] = rb::f32_to_bf16(acc[g][e8 + e] * inv_l);
            }
          }
        }
      }
    }
    return;
  }

  // Merge across the 4 waves via LDS. Layout per (wave, head):
  // [DH acc][m][l] floats.
  __shared__ float lds[HS ? 1 : NW][HS ? 1 : G][DH + 2];
  if (grp == 0) {   // lanes 0..GL-1 of each wave hold the wave's merged state
#pragma unroll
    for (int g = 0; g < GW; ++g) {
#pragma unroll
      for (int e = 0; e < VE; ++e) lds[wid][g][d0 + e] = acc[g][e];
      if (gl == 0) { lds[wid][g][DH] = m[g]; lds[wid][g][DH + 1] = l[g]; }
    }
  }
  __syncthreads();

  // Wave `wid` finalizes heads g = wid, wid+NW, ... (lanes 0..15 active).
  for (int g = wid; g < G; g += NW) {
    if (grp != 0) continue;
    float mm = -INFINITY;
#pragma unroll
    for (int w = 0; w < NW; ++w) mm = fmaxf(mm, lds[w][g][DH]);
    float ll = 0.f, av[VE];
#pragma unroll
    for (int e = 0; e < VE; ++e) av[e] = 0.f;
    if (mm != -INFINITY) {
#pragma unroll
      for (int w = 0; w < NW; ++w) {
        const float a = __expf(lds[w][g][DH] - mm);
        ll += lds[w][g][DH + 1] * a;
#pragma unroll
        for (int e = 0; e < VE; ++e) av[e] += lds[w][g][d0 + e] * a;
      }
    }
    if (SPLIT) {
      float *pp = partial + (((int64_t)b * Hq + hq0 + g) * nsplit + split) * (DH + 2);
#pragma unroll
      for (int e = 0; e < VE; ++e) pp[d0 + e] = av[e];
      if (gl == 0) { pp[DH] = mm; pp[DH + 1] = ll; }
    } else {
      const float inv_l = (ll > 0.f) ? 1.0f / ll : 0.f;
      uint16_t *op = out + ((int64_t)b * Hq + hq0 + g) * DH + d0;
#pragma unroll
      for (int e = 0; e < VE; ++e) op[e] = rb::f32_to_bf16(av[e] * inv_l);
      if (out_swz != nullptr) {
        // also emit the decode-GEMM operand layout for o_proj
        // ([K/16][2][32][8] over K = Hq*DH, m = b): saves the standalone
        // decode_swizzle_x launch per layer. VE >= 8, d0 8-aligned.
#pragma unroll
        for (int e8 = 0; e8 < VE; e8 += 8) {
          const int kk = (hq0 + g) * DH + d0 + e8;
          uint16_t *sp = out_swz + (kk >> 4) * 512 + ((kk >> 3) & 1) * 256 +
                         b * 8;
#pragma unroll
          for (int e = 0; e < 8; ++e)
            sp[e] = rb::f32_to_bf16(av[e8 + e] * inv_l);
        }
      }
    }
  }
}

// Second pass for SPLIT mode: merge nsplit partials per (b, hq) row.
__global__ void decode_combine_kernel(const float *__restrict__ partial,
                                      uint16_t *__restrict__ out,
                                      uint16_t *__restrict__ out_swz,
                                      int Hq, int nsplit,
                                      int dh) {
  const int64_t row = blockIdx.x;          // b * Hq + hq
  const float *p = partial + row * (int64_t)nsplit * (dh + 2);
  float mm = -INFINITY;
  for (int s = 0; s < nsplit; ++s) mm = fmaxf(mm, p[s * (dh + 2) + dh]);
  float ll = 0.f;
  for (int s = 0; s < nsplit; ++s)
    if (p[s * (dh + 2) + dh] != -INFINITY)
      ll += p[s * (dh + 2) + dh + 1] * __expf(p[s * (dh + 2) + dh] - mm);
  const float inv_l = (ll > 0.f) ? 1.0f / ll : 0.f;
  const int b = (int)(row / Hq);
  const int h = (int)(row % Hq);
  for (int d = threadIdx.x; d < dh; d += blockDim.x) {
    float o = 0.f;
    for (int s = 0; s < nsplit; ++s) {
      const float ms = p[s * (dh + 2) + dh];
      if (ms != -INFINITY) o += p[s * (dh + 2) + d] * __expf(ms - mm);
    }
    const uint16_t ob = rb::f32_to_bf16(o * inv_l);
    out[row * dh + d] = ob;
    if (out_swz != nullptr) {
      const int kk = h * dh + d;
      out_swz[(kk >> 4) * 512 + ((kk >> 3) & 1) * 256 + b * 8 +
              (kk & 7)] = ob;
    }
  }
}

// ---------------------------------------------------------------------------
// MFMA decode for GQA/MQA (G >= 4), bf16 cache, DH in {64, 128}, bs == 16.
//
// The scalar formulation above spends ~30 VALU ops per (token, head) on
// dot -> shfl-reduce -> exp -> acc; at G >= 4 that serial chain is the
// bind (PMC: 57% WAIT_ANY / 28% VALU at 2 waves/SIMD pre-head-split,
// and the head-split variant re-reads K/V NW times through L2). Here
// the score block S[16 tok][G] is ONE mfma_f32_16x16x32_bf16 chain over
// DH, and O^T accumulates via mfma_f32_32x32x16_bf16 — each K/V row is
// read once, and per-block VALU drops ~60x.
//
//  * wave owns whole 16-token cache blocks (bi = blk_lo + wid, step NW).
//  * K is read DIRECTLY in A-fragment order from the [BS, Dh] block
//    layout: lane l reads K[token l&15][dh (l>>4)*8 + 32*ks] — 16 B per
//    lane from 16 adjacent 256 B rows inside one contiguous 4 KB block
//    (64 B segments, L2-friendly), no staging.
//  * V comes from the TRANSPOSED cache layout ([Dh, BS] per block,
//    alloc_kv_cache v_transposed — the framework owns the layout, so
//    the PV A-fragment is a direct 16 B/lane read too. The first cut
//    staged the transpose through per-wave LDS; PMC showed 26-43% of
//    wave cycles burned in LDS bank conflicts (the b16 scatter's 8-row
//    lane stride lands on 2 of 64 banks at any 16-aligned row pitch).
//  * fragment maps (guide + mfma_probe_16x16x32 GPU test):
//      16x16x32: A[i=l&15][k=(l>>4)*8+e]; B[k][j=l&15]; C[j=l&15][i=(l>>4)*4+r]
//      32x32x16: A[i=l&31][k=(l>>5)*8+e]; C[j=l&31][i=(r&3)+8*(r>>2)+4*(l>>5)]
// ---------------------------------------------------------------------------
// FP8 mode: K rows stay the scalar-path e4m3 layout ([BS][DH bytes +
// f32 scale + pad], fp8_row_bytes); V blocks are TRANSPOSED e4m3
// ([DH+4][BS] bytes — the 4 tail "rows" are 64 B = the 16 per-token
// f32 scales). Fragments are 8 B/lane raw loads dequanted at use; the
// per-token K scale folds into the S rows after the score MFMA and the
// V scale into P before the PV MFMA (softmax l/m stay on unscaled P).
template <int DH, int G, bool SPLIT, bool FP8 = false>
__global__ __launch_bounds__(BLOCK) void paged_decode_mfma_kernel(
    const uint16_t *__restrict__ q, const uint16_t *__restrict__ k_cache,
    const uint16_t *__restrict__ v_cache,
    const int32_t *__restrict__ block_tables,
    const int32_t *__restrict__ seq_lens,
    const int32_t *__restrict__ seq_starts, uint16_t *__restrict__ out,
    uint16_t *__restrict__ out_swz, float *__restrict__ partial, int hkv,
    int max_blocks, int nsplit, float scale) {
  constexpr int BS = 16;                  // cache block = one S tile
  constexpr int KS32 = DH / 32;           // QK^T contraction steps
  constexpr int DT = DH / 32;             // O^T 32-row d-tiles
  constexpr int FRW = FP8 ? 2 : 4;        // u32 words per raw fragment

  const int b = blockIdx.x;
  const int h_kv = blockIdx.y;
  const int split = SPLIT ? blockIdx.z : 0;
  const int hq0 = h_kv * G;
  const int Hq = hkv * G;

  const int seq_len = seq_lens[b];
  const int start = (seq_starts != nullptr) ? seq_starts[b] : 0;
  int t_begin = start, t_end = seq_len;
  if (SPLIT) {
    const int len = seq_len - start;
    const int chunk = (len + nsplit - 1) / nsplit;
    t_begin = start + split * chunk;
    t_end = min(seq_len, t_begin + chunk);
  }

  const int tid = (int)threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int i16 = lane & 15;
  const int h16 = lane >> 4;              // 0..3
  const int h32 = lane >> 5;              // 0..1
  const int c32 = lane & 31;

  const int32_t *bt = block_tables + (int64_t)b * max_blocks;

  // block-table staging (shared by all waves; same scheme as scalar)
  constexpr int BT_LDS = 512;
  __shared__ int32_t bt_lds[BT_LDS];
  const int blk0 = t_begin / BS;
  if (t_begin < t_end) {
    const int nbt = min((t_end - 1) / BS - blk0 + 1, BT_LDS);
    for (int i = tid; i < nbt; i += BLOCK) bt_lds[i] = bt[blk0 + i];
  }
  // cross-wave merge region
  __shared__ float mlds[NW][G][DH + 2];
  __syncthreads();

  // Q fragments (B of the swapped QK^T), pre-scaled; head = i16.
  rb_bf16x8v qf[KS32];
  {
    const bool hv = i16 < G;
    const uint16_t *qp =
        q + ((int64_t)b * Hq + hq0 + (hv ? i16 : 0)) * DH;
#pragma unroll
    for (int ks = 0; ks < KS32; ++ks) {
      float f[8];
      rb::VIO<uint16_t>::load(qp + ks * 32 + h16 * 8, f);
      union { unsigned short u[8]; rb_bf16x8v v; } c;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        c.u[e] = rb::f32_to_bf16(hv ? f[e] * scale : 0.0f);
      qf[ks] = c.v;
    }
  }

  rb_f32x16v acc_o[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) acc_o[dt] = (rb_f32x16v)(0.0f);
  float m_run = -INFINITY, l_run = 0.0f;

  const int blk_lo = t_begin / BS;
  const int blk_hi = (t_end > t_begin) ? (t_end - 1) / BS : blk_lo - 1;

  // 2-deep block ring: iteration i+1's K/V raw fragments are issued
  // while i computes (PMC post-VT-layout: 74% WAIT_ANY at occupancy 2 —
  // two waves/SIMD cannot hide HBM latency without in-flight depth).
  // +(KS32+DT)*FRW words per slot keeps occupancy at 2.
  uint32_t kfr[2][KS32][FRW], vfr[2][DT][FRW];
  float ksc[2][4], vsc[2][4];            // fp8 per-token scales

  auto fetch = [&](int bi, int slot) {
    const int bi_l = bi - blk0;
    const int blk = (bi_l < BT_LDS) ? bt_lds[bi_l] : bt[bi];
    if (FP8) {
      constexpr int RB8 = DH + 16;       // fp8 K row bytes
      const uint8_t *kb = reinterpret_cast<const uint8_t *>(k_cache) +
          ((int64_t)blk * hkv + h_kv) * BS * RB8;
      const uint8_t *vb = reinterpret_cast<const uint8_t *>(v_cache) +
          ((int64_t)blk * hkv + h_kv) * (DH + 4) * BS;
#pragma unroll
      for (int ks = 0; ks < KS32; ++ks)
        rb::ld_words<2>(kfr[slot][ks], kb + i16 * RB8 + ks * 32 + h16 * 8);
#pragma unroll
      for (int dt = 0; dt < DT; ++dt)
        rb::ld_words<2>(vfr[slot][dt], vb + (dt * 32 + c32) * BS + h32 * 8);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int tok = h16 * 4 + r;
        ksc[slot][r] = *reinterpret_cast<const float *>(kb + tok * RB8 + DH);
        vsc[slot][r] = *reinterpret_cast<const float *>(
            vb + DH * BS + tok * 4);
      }
    } else {
      const uint16_t *kb = k_cache + ((int64_t)blk * hkv + h_kv) * BS * DH;
      // v block is [DH][BS] (transposed layout)
      const uint16_t *vb = v_cache + ((int64_t)blk * hkv + h_kv) * DH * BS;
#pragma unroll
      for (int ks = 0; ks < KS32; ++ks)
        rb::ld_words<4>(kfr[slot][ks], reinterpret_cast<const uint8_t *>(
            kb + i16 * DH + ks * 32 + h16 * 8));
#pragma unroll
      for (int dt = 0; dt < DT; ++dt)
        rb::ld_words<4>(vfr[slot][dt], reinterpret_cast<const uint8_t *>(
            vb + (dt * 32 + c32) * BS + h32 * 8));
    }
  };

  // raw fragment -> bf16 MFMA operand (fp8: dequant UNSCALED — scales
  // fold into S / P outside the MFMA)
  auto frag = [&](const uint32_t *raw) -> rb_bf16x8v {
    union { unsigned u[4]; rb_bf16x8v v; } c;
    if (FP8) {
      float f[8];
      rb::fp8w_to_f32(raw[0], f);
      rb::fp8w_to_f32(raw[1], f + 4);
#pragma unroll
      for (int e = 0; e < 4; ++e)
        c.u[e] = rb_pack_bf16(f[2 * e], f[2 * e + 1]);
    } else {
      c.u[0] = raw[0]; c.u[1] = raw[1]; c.u[2] = raw[2]; c.u[3] = raw[3];
    }
    return c.v;
  };

  auto step = [&](int bi, int cur) {
    if (bi + NW <= blk_hi) fetch(bi + NW, cur ^ 1);

    // ---- S[16 tok][16 heads] = K_blk @ Q^T ----------------------------
    rb_f32x4v s4 = (rb_f32x4v)(0.0f);
#pragma unroll
    for (int ks = 0; ks < KS32; ++ks) {
      s4 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(frag(kfr[cur][ks]),
                                                   qf[ks], s4, 0, 0, 0);
    }
    if (FP8) {
#pragma unroll
      for (int r = 0; r < 4; ++r) s4[r] *= ksc[cur][r];
    }

    // ---- mask + online softmax (state per head = per col = per lane) --
    float p4[4];
    float mx = -INFINITY;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int t = bi * BS + h16 * 4 + r;
      p4[r] = (t >= t_begin && t < t_end) ? s4[r] : -INFINITY;
      mx = fmaxf(mx, p4[r]);
    }
    mx = fmaxf(mx, __shfl_xor(mx, 16, 64));
    mx = fmaxf(mx, __shfl_xor(mx, 32, 64));
    const float mn = fmaxf(m_run, mx);
    if (mn != -INFINITY) {
      const float alpha = (m_run == -INFINITY) ? 0.0f : __expf(m_run - mn);
      float psum = 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        p4[r] = (p4[r] == -INFINITY) ? 0.0f : __expf(p4[r] - mn);
        psum += p4[r];
      }
      psum += __shfl_xor(psum, 16, 64);
      psum += __shfl_xor(psum, 32, 64);
      l_run = l_run * alpha + psum;
      m_run = mn;
      if (alpha != 1.0f) {
#pragma unroll
        for (int dt = 0; dt < DT; ++dt)
#pragma unroll
          for (int r = 0; r < 16; ++r) acc_o[dt][r] *= alpha;
      }

      // ---- P[tok][head] -> B fragment of PV (k = tok, col = head) ----
      // source lane s = head + 16*(tok>>2) holds tokens 4*(s>>4)+r.
      // target lane l needs tokens (l>>5)*8..+8 of head l&31 (<16).
      // fp8: the per-token V scale rides on P here (l/m and psum above
      // stay on the unscaled softmax numerators).
      float pv0 = p4[0], pv1 = p4[1], pv2 = p4[2], pv3 = p4[3];
      if (FP8) {
        pv0 *= vsc[cur][0]; pv1 *= vsc[cur][1];
        pv2 *= vsc[cur][2]; pv3 *= vsc[cur][3];
      }
      const unsigned w01 = rb_pack_bf16(pv0, pv1);
      const unsigned w23 = rb_pack_bf16(pv2, pv3);
      const int sh = c32 & 15;            // source head (cols 16+ unused)
      const int s1 = sh + 16 * (h32 * 2);
      const int s2 = s1 + 16;
      union { unsigned u[4]; rb_bf16x8v v; } pf;
      pf.u[0] = __shfl(w01, s1, 64);
      pf.u[1] = __shfl(w23, s1, 64);
      pf.u[2] = __shfl(w01, s2, 64);
      pf.u[3] = __shfl(w23, s2, 64);

      // ---- O^T[d][head] += V^T_blk @ P --------------------------------
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        acc_o[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            frag(vfr[cur][dt]), pf.v, acc_o[dt], 0, 0, 0);
      }
    }
  };

  const int bi0 = blk_lo + wid;
  if (bi0 <= blk_hi) fetch(bi0, 0);
  for (int bi = bi0; bi <= blk_hi; bi += 2 * NW) {
    step(bi, 0);
    if (bi + NW <= blk_hi) step(bi + NW, 1);
  }

  // ---- cross-wave merge via LDS (same scheme as the scalar kernel) ----
  // acc_o lane map: head = c32, d = dt*32 + (r&3)+8*(r>>2)+4*h32.
  if (c32 < G) {
#pragma unroll
    for (int dt = 0; dt < DT; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r)
        mlds[wid][c32][dt * 32 + (r & 3) + 8 * (r >> 2) + 4 * h32] =
            acc_o[dt][r];
  }
  if (lane < 16 && i16 < G) {
    mlds[wid][i16][DH] = m_run;
    mlds[wid][i16][DH + 1] = l_run;
  }
  __syncthreads();

  constexpr int VE = DH / 16;             // elems per merge lane
  const int gl = lane & 15;
  const int d0 = gl * VE;
  for (int g = wid; g < G; g += NW) {
    if (lane >= 16) continue;
    float mm = -INFINITY;
#pragma unroll
    for (int w = 0; w < NW; ++w) mm = fmaxf(mm, mlds[w][g][DH]);
    float ll = 0.f, av[VE];
#pragma unroll
    for (int e = 0; e < VE; ++e) av[e] = 0.f;
    if (mm != -INFINITY) {
#pragma unroll
      for (int w = 0; w < NW; ++w) {
        const float a = __expf(mlds[w][g][DH] - mm);
        ll += mlds[w][g][DH + 1] * a;
#pragma unroll
        for (int e = 0; e < VE; ++e) av[e] += mlds[w][g][d0 + e] * a;
      }
    }
    if (SPLIT) {
      float *pp =
          partial + (((int64_t)b * Hq + hq0 + g) * nsplit + split) * (DH + 2);
#pragma unroll
      for (int e = 0; e < VE; ++e) pp[d0 + e] = av[e];
      if (gl == 0) { pp[DH] = mm; pp[DH + 1] = ll; }
    } else {
      const float inv_l = (ll > 0.f) ? 1.0f / ll : 0.f;
      uint16_t *op = out + ((int64_t)b * Hq + hq0 + g) * DH + d0;
#pragma unroll
      for (int e = 0; e < VE; ++e) op[e] = rb::f32_to_bf16(av[e] * inv_l);
      if (VE >= 8 && out_swz != nullptr) {
#pragma unroll
        for (int e8 = 0; e8 < VE; e8 += 8) {
          const int kk = (hq0 + g) * DH + d0 + e8;
          uint16_t *sp = out_swz + (kk >> 4) * 512 + ((kk >> 3) & 1) * 256 +
                         b * 8;
#pragma unroll
          for (int e = 0; e < 8 && e < VE; ++e)
            sp[e] = rb::f32_to_bf16(av[e8 + e] * inv_l);
        }
      }
    }
  }
}

template <int DH, int G, bool FP8>
void launch_decode(const at::Tensor &q, const at::Tensor &k_cache,
                   const at::Tensor &v_cache, const at::Tensor &block_tables,
                   const at::Tensor &seq_lens,
                   const c10::optional<at::Tensor> &seq_starts,
                   at::Tensor &out, const c10::optional<at::Tensor> &out_swz,
                   int nsplit,
                   float scale, hipStream_t stream) {
  const int32_t *starts = seq_starts.has_value()
      ? seq_starts->data_ptr<int32_t>() : nullptr;
  const int B = (int)q.size(0);
  const int hkv = (int)k_cache.size(1);
  const int bs = (int)k_cache.size(2);
  const int max_blocks = (int)block_tables.size(1);
  uint16_t *swz = out_swz.has_value() ? (uint16_t *)out_swz->data_ptr()
                                      : nullptr;
  // MFMA route: keyed on the TRANSPOSED V layout (bf16 [Dh, BS] /
  // fp8 [Dh+4, BS] blocks) the allocator chooses for MFMA-eligible
  // models (G >= 4, DH <= 128, 16-token blocks — RB_DECODE_MFMA=0 at
  // alloc reverts). The layout is the single source of truth: a
  // transposed cache can only be read by the MFMA kernel and vice
  // versa.
  const bool vt = v_cache.size(3) == bs &&
      v_cache.size(2) == (int64_t)(FP8 ? DH + 4 : DH);
  if constexpr (G >= 4 && DH <= 128) {
    if (vt && bs == 16) {
      if (nsplit <= 1) {
        hipLaunchKernelGGL((paged_decode_mfma_kernel<DH, G, false, FP8>),
                           dim3(B, hkv, 1), dim3(BLOCK), 0, stream,
                           (const uint16_t *)q.data_ptr(),
                           (const uint16_t *)k_cache.data_ptr(),
                           (const uint16_t *)v_cache.data_ptr(),
                           block_tables.data_ptr<int32_t>(),
                           seq_lens.data_ptr<int32_t>(), starts,
                           (uint16_t *)out.data_ptr(), swz, nullptr, hkv,
                           max_blocks, 1, scale);
      } else {
        const int Hq = hkv * G;
        auto partial = at::empty({B, Hq, nsplit, DH + 2},
                                 q.options().dtype(at::kFloat));
        hipLaunchKernelGGL((paged_decode_mfma_kernel<DH, G, true, FP8>),
                           dim3(B, hkv, nsplit), dim3(BLOCK), 0, stream,
                           (const uint16_t *)q.data_ptr(),
                           (const uint16_t *)k_cache.data_ptr(),
                           (const uint16_t *)v_cache.data_ptr(),
                           block_tables.data_ptr<int32_t>(),
                           seq_lens.data_ptr<int32_t>(), starts,
                           nullptr, nullptr, partial.data_ptr<float>(), hkv,
                           max_blocks, nsplit, scale);
        hipLaunchKernelGGL(decode_combine_kernel, dim3(B * Hq), dim3(256), 0,
                           stream, partial.data_ptr<float>(),
                           (uint16_t *)out.data_ptr(), swz, Hq, nsplit, DH);
      }
      return;
    }
  }
  TORCH_CHECK(!vt, "paged_decode: transposed V cache requires the MFMA "
              "path (bf16, G>=4, Dh<=128, bs==16); got G=", G, " Dh=", DH);
  if (nsplit <= 1) {
    hipLaunchKernelGGL((paged_decode_kernel<DH, G, false, FP8>),
                       dim3(B, hkv, 1),
                       dim3(BLOCK), 0, stream, (const uint16_t *)q.data_ptr(),
                       (const uint16_t *)k_cache.data_ptr(),
                       (const uint16_t *)v_cache.data_ptr(),
                       block_tables.data_ptr<int32_t>(), seq_lens.data_ptr<int32_t>(),
                       starts,
                       (uint16_t *)out.data_ptr(), swz, nullptr, hkv, bs,
                       max_blocks, 1, scale);
  } else {
    const int Hq = hkv * G;
    auto partial = at::empty({B, Hq, nsplit, DH + 2},
                             q.options().dtype(at::kFloat));
    hipLaunchKernelGGL((paged_decode_kernel<DH, G, true, FP8>),
                       dim3(B, hkv, nsplit),
                       dim3(BLOCK), 0, stream, (const uint16_t *)q.data_ptr(),
                       (const uint16_t *)k_cache.data_ptr(),
                       (const uint16_t *)v_cache.data_ptr(),
                       block_tables.data_ptr<int32_t>(), seq_lens.data_ptr<int32_t>(),
                       starts,
                       nullptr, nullptr, partial.data_ptr<float>(), hkv, bs,
                       max_blocks, nsplit, scale);
    hipLaunchKernelGGL(decode_combine_kernel, dim3(B * Hq), dim3(256), 0, stream,
                       partial.data_ptr<float>(), (uint16_t *)out.data_ptr(),
                       swz, Hq, nsplit, DH);
  }
}

}  // namespace

std::vector<at::Tensor> paged_decode_swz(at::Tensor q, at::Tensor k_cache,
                                         at::Tensor v_cache,
                                         at::Tensor block_tables,
                                         at::Tensor seq_lens, int64_t nsplit,
                                         double scale,
                                         c10::optional<at::Tensor> seq_starts);

at::Tensor paged_decode(at::Tensor q, at::Tensor k_cache, at::Tensor v_cache,
                        at::Tensor block_tables, at::Tensor seq_lens,
                        int64_t nsplit, double scale,
                        c10::optional<at::Tensor> seq_starts) {
  return paged_decode_swz(q, k_cache, v_cache, block_tables, seq_lens,
                          nsplit, scale, seq_starts)[0];
}

// Variant also returning the decode-GEMM operand layout of the output
// (o_proj input swizzle fused into the attention epilogue).
std::vector<at::Tensor> paged_decode_swz(at::Tensor q, at::Tensor k_cache,
                                         at::Tensor v_cache,
                                         at::Tensor block_tables,
                                         at::Tensor seq_lens, int64_t nsplit,
                                         double scale,
                                         c10::optional<at::Tensor> seq_starts) {
  if (seq_starts.has_value()) {
    TORCH_CHECK(seq_starts->scalar_type() == at::kInt &&
                seq_starts->is_contiguous(), "paged_decode: seq_starts");
  }
  TORCH_CHECK(q.is_cuda() && q.is_contiguous(), "paged_decode: q");
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "paged_decode: bf16 only");
  TORCH_CHECK(block_tables.scalar_type() == at::kInt && seq_lens.scalar_type() == at::kInt,
              "paged_decode: int32 metadata");
  const int Hq = (int)q.size(1);
  const int dh = (int)q.size(2);
  const int hkv = (int)k_cache.size(1);
  TORCH_CHECK(Hq % hkv == 0, "paged_decode: Hq % Hkv");
  const int G = Hq / hkv;
  auto out = at::empty_like(q);
  // emit the o_proj operand layout only for decode-batch shapes
  c10::optional<at::Tensor> out_swz;
  const int B0 = (int)q.size(0);
  if (B0 <= 32 && (Hq * dh) % 16 == 0) {
    out_swz = at::empty({(int64_t)(Hq * dh / 16) * 512}, q.options());
  }
  auto stream = at::hip::getCurrentHIPStream();

  const bool fp8 = k_cache.scalar_type() == at::kByte;
  if (fp8) {
    TORCH_CHECK((int)k_cache.size(3) == dh + 16 &&
                v_cache.scalar_type() == at::kByte,
                "paged_decode: fp8 cache rows must be dh+16 bytes");
  }
#define RB_DEC(DHV, GV)                                                        \
  do {                                                                         \
    if (fp8)                                                                   \
      launch_decode<DHV, GV, true>(q, k_cache, v_cache, block_tables,         \
                                   seq_lens, seq_starts, out, out_swz,        \
                                   (int)nsplit, (float)scale, stream);        \
    else                                                                       \
      launch_decode<DHV, GV, false>(q, k_cache, v_cache, block_tables,        \
                                    seq_lens, seq_starts, out, out_swz,       \
                                    (int)nsplit, (float)scale, stream);       \
  } while (0)
  if (dh == 128) {
    switch (G) {
      case 1: RB_DEC(128, 1); break;
      case 2: RB_DEC(128, 2); break;
      case 4: RB_DEC(128, 4); break;
      case 8: RB_DEC(128, 8); break;
      default: TORCH_CHECK(false, "paged_decode: unsupported G=", G);
    }
  } else if (dh == 64) {
    switch (G) {
      case 1: RB_DEC(64, 1); break;
      case 2: RB_DEC(64, 2); break;
      case 4: RB_DEC(64, 4); break;
      case 8: RB_DEC(64, 8); break;
      case 16: RB_DEC(64, 16); break;
      default: TORCH_CHECK(false, "paged_decode: unsupported G=", G);
    }
  } else if (dh == 256) {
    // gemma-7b (wide heads, MHA). VE=16 -> ~70 VGPRs at G=1; first GPU
    // validation is the RB_EXPERIMENTAL test (round 2).
    switch (G) {
      case 1: RB_DEC(256, 1); break;
      case 2: RB_DEC(256, 2); break;
      default: TORCH_CHECK(false, "paged_decode: unsupported G=", G);
    }
  } else {
    TORCH_CHECK(false, "paged_decode: Dh must be 64/128/256, got ", dh);
  }
#undef RB_DEC
  if (out_swz.has_value()) return {out, *out_swz};
  return {out};
}
