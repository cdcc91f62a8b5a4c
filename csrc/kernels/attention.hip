// Hand-written CDNA4 (gfx950) flash attention, head_dim = 64, bf16.
//
// Replaces the AOTriton SDPA kernels on the BERT/GPT hot path (measured
// ~315 TF fwd / ~183 TF bwd on MI355X at our shapes; see profiles/).
//
// Forward structure (cdna_hip_programming.md section B recipe, adapted to
// one 32-row q-block per wave, 32-key kv tiles, head_dim 64):
//  * swapped QK^T: S^T = K @ Q^T via v_mfma_f32_32x32x16_bf16, so each
//    lane owns ONE q-row's scores (16 kv values per lane, the partner
//    lane l^32 holds the other 16) -> online softmax is 15 reg-max ops +
//    one shfl_xor(32) exchange, no LDS.
//  * O accumulated TRANSPOSED: O^T = V^T @ P^T, so the online rescale by
//    alpha = exp(m_old - m_new) is a lane-local scalar multiply.
//  * P^T fragments assembled in-register: pack f32 pairs to bf16 dwords
//    and exchange halves with v_permlane32_swap (T12).
//  * Q/K fragments are contiguous 16-byte row segments loaded straight
//    from global (L2-resident tiles); V^T fragments gather columns (L2).
// Saves per-row LSE (m + log l) for the backward.
//
// Backward: two kernels (kv-parallel for dK/dV, q-parallel for dQ), each
// lane owning its output row — no atomics.  D_i = rowsum(dO*O) is
// precomputed by a small kernel.  All kernels take (batch, head, seq)
// strides so the qkv-unbind views are consumed without copies.

#include <hip/hip_runtime.h>
#include <cstdint>

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using bf16x4 = __attribute__((ext_vector_type(4))) short;
using f32x16 = __attribute__((ext_vector_type(16))) float;
using i32x2 = __attribute__((ext_vector_type(2))) int;

#define DEVI __device__ __forceinline__

namespace {

DEVI float bf2f(short u) {
  union { unsigned int i; float f; } v;
  v.i = ((unsigned int)(unsigned short)u) << 16;
  return v.f;
}

DEVI unsigned short f2bf(float f) {
  union { float f; unsigned int i; } v;
  v.f = f;
  unsigned int lsb = (v.i >> 16) & 1u;
  return (unsigned short)((v.i + 0x7fffu + lsb) >> 16);
}

DEVI unsigned int pack2(float a, float b) {
  return (unsigned int)f2bf(a) | ((unsigned int)f2bf(b) << 16);
}

// C/D row for reg r, half hi of v_mfma_f32_32x32x16_bf16
DEVI int crow(int r, int hi) {
  return (r & 3) + 8 * (r >> 2) + 4 * hi;
}

// assemble the P^T mfma operand from 8 f32 scores (p[0..7] at rows
// crow(0..7, hi)): returns bf16x8 with element j = P at kv = hi*8+j
DEVI bf16x8 assemble_pfrag(const float* p) {
  unsigned int c0 = pack2(p[0], p[1]);
  unsigned int c1 = pack2(p[2], p[3]);
  unsigned int c2 = pack2(p[4], p[5]);
  unsigned int c3 = pack2(p[6], p[7]);
  i32x2 r = __builtin_amdgcn_permlane32_swap((int)c0, (int)c2, false, false);
  i32x2 s = __builtin_amdgcn_permlane32_swap((int)c1, (int)c3, false, false);
  union { unsigned int d[4]; bf16x8 v; } out;
  out.d[0] = (unsigned int)r[0];
  out.d[1] = (unsigned int)s[0];
  out.d[2] = (unsigned int)r[1];
  out.d[3] = (unsigned int)s[1];
  return out.v;
}

// ----------------------------------------------------------------------------
// philox4x32-10 counter-based RNG for in-kernel attention dropout.
// One call yields 16 bytes = the 16 Bernoulli draws of one lane's
// 32-key sub-tile (byte u >= thresh keeps; p is quantized to 1/256 —
// documented in ops/attention.py).  The keep-mask is packed into a
// [bh, seq, ceil(seq/32)] uint32 tensor during the FORWARD; the three
// backward kernels re-read bits instead of re-running philox.
// ----------------------------------------------------------------------------
DEVI void philox_round(unsigned int& c0, unsigned int& c1,
                       unsigned int& c2, unsigned int& c3,
                       unsigned int k0, unsigned int k1) {
  const unsigned int hi0 = __umulhi(0xD2511F53u, c0);
  const unsigned int lo0 = 0xD2511F53u * c0;
  const unsigned int hi1 = __umulhi(0xCD9E8D57u, c2);
  const unsigned int lo1 = 0xCD9E8D57u * c2;
  c0 = hi1 ^ c1 ^ k0;
  c1 = lo1;
  c2 = hi0 ^ c3 ^ k1;
  c3 = lo0;
}

DEVI void philox4x32(unsigned int c0, unsigned int c1, unsigned int c2,
                     unsigned int c3, unsigned int k0, unsigned int k1,
                     unsigned int out[4]) {
#pragma unroll
  for (int i = 0; i < 10; ++i) {
    philox_round(c0, c1, c2, c3, k0, k1);
    k0 += 0x9E3779B9u;
    k1 += 0xBB67AE85u;
  }
  out[0] = c0; out[1] = c1; out[2] = c2; out[3] = c3;
}

// ============================================================================
// forward (v3): KVBLK = 64; K (row copies) and V^T (transposed, 144-
// byte row pad -> conflict-free ds_read_b128 fragments) staged in LDS
// once per block per tile.  Measured: explicit double-buffering and
// register prefetch both REGRESS (hipcc schedules the simple form best
// at 3 waves/SIMD) - keep this structure.
//
// v3: the kv loop is SPLIT into a branch-free bulk phase (tiles that are
// provably full: no seq-bound or causal masking, no per-wave break — the
// compiler pipelines it like the non-causal fast path) and a masked
// phase covering the causal diagonal (<= 2 tiles) and the seq tail.
// Round-1 measurement showed the single masked loop ran EVERY causal
// tile 33-46% slower per tile than the non-causal loop ran the same
// machine code — the masking ALU + dynamic break cost lands on all
// tiles, not just diagonal ones.
// ============================================================================

// one 32-key sub-tile of the fwd online-softmax loop; MASKED adds the
// seq-bound + causal-diagonal masking (bulk tiles skip all of it);
// DROP generates + applies + publishes the dropout keep-mask (the
// normalizer l accumulates UNDROPPED exp values, so lse and the
// backward's P are dropout-free; O accumulates P*keep/(1-p)*V)
template <bool MASKED, bool DROP, int D>
DEVI void fwd_tile(const short (&ldsK)[64][D + 8],
                   const short (&ldsV)[D][72],
                   int sub, int64_t kvs, int64_t seq, int64_t myq,
                   float scale, int causal, const bf16x8 (&qfrag)[D / 16],
                   f32x16 (&ot)[D / 32], float& m, float& l, int lq,
                   int hi, unsigned int* maskrow, int64_t mask_w,
                   unsigned int seed0, unsigned int seed1,
                   unsigned int bh32, int drop_thresh, float inv_keep) {
  f32x16 st = {};
#pragma unroll
  for (int c = 0; c < D / 16; ++c) {
    bf16x8 kfrag =
        *reinterpret_cast<const bf16x8*>(&ldsK[sub + lq][hi * 8 + 16 * c]);
    st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfrag, qfrag[c], st, 0, 0,
                                                 0);
  }
  float s[16];
  float tile_max = -1e30f;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    float sv = st[r] * scale;
    if (MASKED) {
      const int64_t kvg = kvs + crow(r, hi);
      if (kvg >= seq || (causal && kvg > myq)) sv = -1e30f;
    }
    s[r] = sv;
    tile_max = fmaxf(tile_max, sv);
  }
  tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32, 64));
  const float m_new = fmaxf(m, tile_max);
  const float alpha = __expf(m - m_new);
  float rowsum = 0.f;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    s[r] = __expf(s[r] - m_new);
    rowsum += s[r];
  }
  rowsum += __shfl_xor(rowsum, 32, 64);
  l = l * alpha + rowsum;
  m = m_new;
  if (DROP) {
    unsigned int rnd[4];
    philox4x32((unsigned int)myq,
               (unsigned int)(kvs >> 5) * 2u + (unsigned int)hi,
               bh32, 0x2E1B2137u, seed0, seed1, rnd);
    unsigned int mybits = 0;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const unsigned int byte = (rnd[r >> 2] >> ((r & 3) * 8)) & 0xFFu;
      const bool keep = (int)byte >= drop_thresh;
      s[r] = keep ? s[r] * inv_keep : 0.f;
      mybits |= (unsigned int)keep << crow(r, hi);
    }
    const unsigned int word = mybits | __shfl_xor(mybits, 32, 64);
    if (hi == 0 && myq < seq)
      maskrow[myq * mask_w + (kvs >> 5)] = word;
  }
#pragma unroll
  for (int j = 0; j < D / 32; ++j)
#pragma unroll
    for (int r = 0; r < 16; ++r)
      ot[j][r] *= alpha;
  bf16x8 pf0 = assemble_pfrag(&s[0]);
  bf16x8 pf1 = assemble_pfrag(&s[8]);
#pragma unroll
  for (int kc = 0; kc < 2; ++kc) {
    bf16x8 pf = kc == 0 ? pf0 : pf1;
#pragma unroll
    for (int j = 0; j < D / 32; ++j) {
      bf16x8 vt = *reinterpret_cast<const bf16x8*>(
          &ldsV[j * 32 + lq][sub + kc * 16 + hi * 8]);
      ot[j] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vt, pf, ot[j], 0, 0,
                                                      0);
    }
  }
}

template <bool DROP, int D>
__global__ __launch_bounds__(256, D == 64 ? 3 : 2) void attn_fwd_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, short* __restrict__ out,
    float* __restrict__ lse, int64_t seq, float scale, int causal,
    int64_t heads, int64_t in_sb, int64_t in_sh, int64_t in_ss,
    int64_t o_sb, int64_t o_sh, int64_t o_ss,
    unsigned int* __restrict__ mask, int64_t mask_w,
    unsigned long long seed, int drop_thresh, float inv_keep) {
  __shared__ short ldsV[2][D][72];     // V^T: [d][kv], double-buffered
  __shared__ short ldsK[2][64][D + 8];  // K: [kv][d]
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;
  // causal: qb rides the Y axis with bh on X — dispatch is x-fastest,
  // so EVERY batch-head of the longest q-block launches before the
  // next-longest (global longest-first packing).  The old per-row
  // reversal still launched one longest block in the LAST row — a
  // ~1-block drain tail that left causal at ~35% packing efficiency
  // at s4096 (profiles/r02_attention_ab.txt wave-residency probe).
  const int64_t bh = causal ? (int64_t)blockIdx.x : (int64_t)blockIdx.y;
  const int64_t boff = (bh / heads) * in_sb + (bh % heads) * in_sh;
  const int64_t qb = causal ? (int64_t)gridDim.y - 1 - blockIdx.y
                            : (int64_t)blockIdx.x;
  const int64_t q0_blk = qb * 128;
  const int64_t q0 = q0_blk + wave * 32;
  const bool active = q0 < seq;
  const short* qp = q + boff;
  const short* kp = k + boff;
  const short* vp = v + boff;

  const int64_t myq = q0 + lq;
  const int64_t qrow = myq < seq ? myq : seq - 1;

  bf16x8 qfrag[D / 16];
#pragma unroll
  for (int c = 0; c < D / 16; ++c)
    qfrag[c] = *reinterpret_cast<const bf16x8*>(
        qp + qrow * in_ss + hi * 8 + 16 * c);

  f32x16 ot[D / 32] = {};
  float m = -1e30f, l = 0.f;

  // bulk tiles are provably full for EVERY lane of the block: below the
  // causal diagonal (kv < q0_blk <= myq) and inside the seq bound
  const int64_t bulk_end = causal ? q0_blk : (seq & ~(int64_t)63);
  const int64_t blk_kv_end =
      causal ? (q0_blk + 128 < seq ? q0_blk + 128 : seq) : seq;
  const int stage_kv = threadIdx.x & 63;
  const int stage_d0 = (threadIdx.x >> 6) * 8;

  // Single-barrier double-buffered staging: the NEXT tile's global
  // loads issue right after the barrier and their latency hides under
  // the CURRENT tile's MFMA/softmax; the ds_write that consumes them
  // (start of the next iteration) is where the vmcnt wait lands.  One
  // __syncthreads per 64-kv tile instead of two (PMC r2: 38% of
  // wave-cycles were parked at barriers).
  bf16x8 pv[D / 32], pk[D / 32];
  auto load_tile = [&](int64_t kv0) {
    int64_t vrow = kv0 + stage_kv;
    if (vrow >= seq) vrow = seq - 1;   // masked columns never contribute
#pragma unroll
    for (int h2 = 0; h2 < D / 32; ++h2) {
      const int sd = stage_d0 + h2 * 32;
      pv[h2] = *reinterpret_cast<const bf16x8*>(vp + vrow * in_ss + sd);
      pk[h2] = *reinterpret_cast<const bf16x8*>(kp + vrow * in_ss + sd);
    }
  };
  auto write_tile = [&](int b) {
#pragma unroll
    for (int h2 = 0; h2 < D / 32; ++h2) {
      const int sd = stage_d0 + h2 * 32;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        ldsV[b][sd + j][stage_kv] = pv[h2][j];
      *reinterpret_cast<bf16x8*>(&ldsK[b][stage_kv][sd]) = pk[h2];
    }
  };

  unsigned int* maskrow =
      DROP ? mask + bh * seq * mask_w : (unsigned int*)nullptr;
  const unsigned int seed0 = (unsigned int)seed;
  const unsigned int seed1 = (unsigned int)(seed >> 32);
  const unsigned int bh32 = (unsigned int)bh;

  int buf = 0;
  if (blk_kv_end > 0) load_tile(0);
  int64_t kv0 = 0;
  for (; kv0 < bulk_end; kv0 += 64) {
    write_tile(buf);
    __syncthreads();
    if (kv0 + 64 < blk_kv_end) load_tile(kv0 + 64);
#pragma unroll
    for (int sub = 0; sub < 64; sub += 32)
      fwd_tile<false, DROP, D>(ldsK[buf], ldsV[buf], sub, kv0 + sub, seq,
                               myq, scale, causal, qfrag, ot, m, l, lq,
                               hi, maskrow, mask_w, seed0, seed1, bh32,
                               drop_thresh, inv_keep);
    buf ^= 1;
  }
  for (; kv0 < blk_kv_end; kv0 += 64) {
    write_tile(buf);
    __syncthreads();
    if (kv0 + 64 < blk_kv_end) load_tile(kv0 + 64);
    if (active) {
      const int64_t wave_kv_end = causal
          ? (q0 + 32 < seq ? q0 + 32 : seq) : seq;
#pragma unroll
      for (int sub = 0; sub < 64; sub += 32) {
        const int64_t kvs = kv0 + sub;
        if (kvs >= wave_kv_end) break;
        fwd_tile<true, DROP, D>(ldsK[buf], ldsV[buf], sub, kvs, seq, myq,
                                scale, causal, qfrag, ot, m, l, lq, hi,
                                maskrow, mask_w, seed0, seed1, bh32,
                                drop_thresh, inv_keep);
      }
    }
    buf ^= 1;
  }

  if (!active || myq >= seq) return;
  const float inv_l = 1.f / l;
  short* op = out + (bh / heads) * o_sb + (bh % heads) * o_sh + myq * o_ss;
#pragma unroll
  for (int j = 0; j < D / 32; ++j)
#pragma unroll
    for (int r = 0; r < 16; ++r)
      op[j * 32 + crow(r, hi)] = (short)f2bf(ot[j][r] * inv_l);
  if (hi == 0) lse[bh * seq + myq] = m + __logf(l);
}

// ============================================================================
// backward: D_i = rowsum(dO * O) precompute
// ============================================================================
// ============================================================================
__global__ void attn_bwd_prep_kernel(const short* __restrict__ dout,
                                     const short* __restrict__ out,
                                     float* __restrict__ delta,
                                     int64_t rows, int64_t seq,
                                     int64_t heads, int64_t do_sb,
                                     int64_t do_sh, int64_t do_ss,
                                     int64_t o_sb, int64_t o_sh,
                                     int64_t o_ss) {
  const int lane = threadIdx.x & 63;
  const int waves_per_block = blockDim.x >> 6;
  for (int64_t row = (int64_t)blockIdx.x * waves_per_block +
                     (threadIdx.x >> 6);
       row < rows; row += (int64_t)gridDim.x * waves_per_block) {
    const int64_t bh = row / seq;
    const int64_t sq = row % seq;
    const short* dp = dout + (bh / heads) * do_sb + (bh % heads) * do_sh +
                      sq * do_ss;
    const short* op = out + (bh / heads) * o_sb + (bh % heads) * o_sh +
                      sq * o_ss;
    float acc = bf2f(dp[lane]) * bf2f(op[lane]);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      acc += __shfl_down(acc, off, 64);
    if (lane == 0) delta[row] = acc;
  }
}

// ============================================================================
// backward kernel A (kv-parallel): dK and dV.  One wave per 32-key kv
// tile, looping q tiles.  Q and dO tiles are staged in LDS as row copies
// once per block: the row fragments read back as conflict-free
// ds_read_b128 and the transposed fragments as pair-broadcast 2-byte LDS
// reads — no global gathers in the loop.
// ============================================================================
__global__ __launch_bounds__(256) void attn_bwd_dkdv_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dk, short* __restrict__ dv, int64_t seq,
    float scale, int causal, int64_t heads, int64_t in_sb, int64_t in_sh,
    int64_t in_ss, int64_t do_sb, int64_t do_sh, int64_t do_ss,
    int64_t g_sb, int64_t g_sh, int64_t g_ss) {
  __shared__ short ldsQ[32][72];
  __shared__ short ldsDO[32][72];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int lkv = lane & 31;
  const int64_t bh = blockIdx.y;
  const int64_t kv0_blk = (int64_t)blockIdx.x * 128;
  const int64_t kv0 = kv0_blk + wave * 32;
  const bool active = kv0 < seq;
  const int64_t boff = (bh / heads) * in_sb + (bh % heads) * in_sh;
  const short* qp = q + boff;
  const short* kp = k + boff;
  const short* vp = v + boff;
  const short* dop = dout + (bh / heads) * do_sb + (bh % heads) * do_sh;
  const float* lsep = lse + bh * seq;
  const float* dltp = delta + bh * seq;

  const int64_t mykv = kv0 + lkv;
  const int64_t kvrow = mykv < seq ? mykv : seq - 1;

  bf16x8 kfrag[4], vfrag[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    kfrag[c] = *reinterpret_cast<const bf16x8*>(
        kp + kvrow * in_ss + hi * 8 + 16 * c);
    vfrag[c] = *reinterpret_cast<const bf16x8*>(
        vp + kvrow * in_ss + hi * 8 + 16 * c);
  }

  f32x16 dvt0 = {}, dvt1 = {};
  f32x16 dkt0 = {}, dkt1 = {};

  const int stage_row = threadIdx.x >> 3;        // 0..31
  const int stage_seg = (threadIdx.x & 7) * 8;   // 0..56
  const int64_t q_start = causal ? (kv0_blk / 32) * 32 : 0;
  for (int64_t q0 = q_start; q0 < seq; q0 += 32) {
    __syncthreads();
    {
      int64_t qr = q0 + stage_row;
      if (qr >= seq) qr = seq - 1;   // masked rows contribute zero
      *reinterpret_cast<bf16x8*>(&ldsQ[stage_row][stage_seg]) =
          *reinterpret_cast<const bf16x8*>(qp + qr * in_ss + stage_seg);
      *reinterpret_cast<bf16x8*>(&ldsDO[stage_row][stage_seg]) =
          *reinterpret_cast<const bf16x8*>(dop + qr * do_ss + stage_seg);
    }
    __syncthreads();
    if (!active || (causal && q0 + 31 < kv0)) continue;
    f32x16 st = {};
    f32x16 dpt = {};
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      bf16x8 qf = *reinterpret_cast<const bf16x8*>(
          &ldsQ[lkv][hi * 8 + 16 * c]);
      bf16x8 dof = *reinterpret_cast<const bf16x8*>(
          &ldsDO[lkv][hi * 8 + 16 * c]);
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf, kfrag[c], st, 0, 0,
                                                   0);
      dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dof, vfrag[c], dpt, 0,
                                                    0, 0);
    }
    float p[16], ds[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int64_t qg = q0 + crow(r, hi);
      const int64_t qgc = qg < seq ? qg : seq - 1;
      float sv = st[r] * scale;
      bool masked = qg >= seq || (causal && mykv > qg) || mykv >= seq;
      float pv = masked ? 0.f : __expf(sv - lsep[qgc]);
      p[r] = pv;
      ds[r] = masked ? 0.f : pv * (dpt[r] - dltp[qgc]) * scale;
    }
    bf16x8 pb0 = assemble_pfrag(&p[0]);
    bf16x8 pb1 = assemble_pfrag(&p[8]);
    bf16x8 db0 = assemble_pfrag(&ds[0]);
    bf16x8 db1 = assemble_pfrag(&ds[8]);
#pragma unroll
    for (int qc = 0; qc < 2; ++qc) {
      bf16x8 dot0, dot1, qt0, qt1;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int qr = qc * 16 + hi * 8 + j;
        dot0[j] = ldsDO[qr][lkv];
        dot1[j] = ldsDO[qr][32 + lkv];
        qt0[j] = ldsQ[qr][lkv];
        qt1[j] = ldsQ[qr][32 + lkv];
      }
      bf16x8 pb = qc == 0 ? pb0 : pb1;
      bf16x8 db = qc == 0 ? db0 : db1;
      dvt0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dot0, pb, dvt0, 0, 0,
                                                     0);
      dvt1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dot1, pb, dvt1, 0, 0,
                                                     0);
      dkt0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qt0, db, dkt0, 0, 0,
                                                     0);
      dkt1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qt1, db, dkt1, 0, 0,
                                                     0);
    }
  }

  if (!active || mykv >= seq) return;
  const int64_t goff = (bh / heads) * g_sb + (bh % heads) * g_sh +
                       mykv * g_ss;
  short* dkp = dk + goff;
  short* dvp = dv + goff;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    dvp[crow(r, hi)] = (short)f2bf(dvt0[r]);
    dvp[32 + crow(r, hi)] = (short)f2bf(dvt1[r]);
    dkp[crow(r, hi)] = (short)f2bf(dkt0[r]);
    dkp[32 + crow(r, hi)] = (short)f2bf(dkt1[r]);
  }
}

// ============================================================================
// split backward A kernels: dV-only and dK-only.  Same structure as the
// combined kernel but half the accumulators each -> 3 waves/SIMD instead
// of 2 (A/B-selectable via epl_attn_bwd mode).  The q loop runs in three
// phases: masked causal-diagonal tiles, branch-free bulk tiles, masked
// seq tail (same rationale as the forward v3 split).
// ============================================================================

// one 32-row q sub-tile (of the 64-row staged chunk) of the dV
// accumulation; `sub` selects the LDS half.  DROP: dV = P_d^T dO, so P
// is masked+rescaled with the forward's keep bits (bit mykv&31 of the
// mask word of row qg — one broadcast word load per row).
template <bool MASKED, bool DROP, int D>
DEVI void dv_tile(const short (&ldsQ)[64][D + 8],
                  const short (&ldsDO)[64][D + 8],
                  int sub, int64_t q0, int64_t seq, int64_t mykv,
                  float scale, int causal, const float* lsep,
                  const bf16x8 (&kfrag)[D / 16], f32x16 (&dvt)[D / 32],
                  int lkv, int hi, const unsigned int* maskrow,
                  int64_t mask_w, float inv_keep) {
  f32x16 st = {};
#pragma unroll
  for (int c = 0; c < D / 16; ++c) {
    bf16x8 qf = *reinterpret_cast<const bf16x8*>(
        &ldsQ[sub + lkv][hi * 8 + 16 * c]);
    st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf, kfrag[c], st, 0, 0, 0);
  }
  float p[16];
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int64_t qg = q0 + crow(r, hi);
    const int64_t qgc = qg < seq ? qg : seq - 1;
    if (MASKED) {
      bool masked = qg >= seq || (causal && mykv > qg) || mykv >= seq;
      p[r] = masked ? 0.f : __expf(st[r] * scale - lsep[qgc]);
    } else {
      p[r] = __expf(st[r] * scale - lsep[qg]);
    }
    if (DROP) {
      const int64_t kvc = mykv < seq ? mykv : seq - 1;
      const unsigned int w = maskrow[qgc * mask_w + (kvc >> 5)];
      p[r] *= ((w >> (kvc & 31)) & 1u) ? inv_keep : 0.f;
    }
  }
  bf16x8 pb0 = assemble_pfrag(&p[0]);
  bf16x8 pb1 = assemble_pfrag(&p[8]);
#pragma unroll
  for (int qc = 0; qc < 2; ++qc) {
    bf16x8 pb = qc == 0 ? pb0 : pb1;
#pragma unroll
    for (int dj = 0; dj < D / 32; ++dj) {
      bf16x8 dot;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int qr = sub + qc * 16 + hi * 8 + j;
        dot[j] = ldsDO[qr][dj * 32 + lkv];
      }
      dvt[dj] =
          __builtin_amdgcn_mfma_f32_32x32x16_bf16(dot, pb, dvt[dj], 0, 0, 0);
    }
  }
}

// one 32-row q sub-tile of the dK accumulation.  DROP mirrors dq_tile:
// P undropped, the dO.V term masked+rescaled.
template <bool MASKED, bool DROP, int D>
DEVI void dk_tile(const short (&ldsQ)[64][D + 8],
                  const short (&ldsDO)[64][D + 8],
                  int sub, int64_t q0, int64_t seq, int64_t mykv,
                  float scale, int causal, const float* lsep,
                  const float* dltp, const bf16x8 (&kfrag)[D / 16],
                  const bf16x8 (&vfrag)[D / 16], f32x16 (&dkt)[D / 32],
                  int lkv, int hi, const unsigned int* maskrow,
                  int64_t mask_w, float inv_keep) {
  f32x16 st = {};
  f32x16 dpt = {};
#pragma unroll
  for (int c = 0; c < D / 16; ++c) {
    bf16x8 qf = *reinterpret_cast<const bf16x8*>(
        &ldsQ[sub + lkv][hi * 8 + 16 * c]);
    bf16x8 dof = *reinterpret_cast<const bf16x8*>(
        &ldsDO[sub + lkv][hi * 8 + 16 * c]);
    st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf, kfrag[c], st, 0, 0, 0);
    dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dof, vfrag[c], dpt, 0, 0,
                                                  0);
  }
  float ds[16];
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int64_t qg = q0 + crow(r, hi);
    const int64_t qgc = qg < seq ? qg : seq - 1;
    float dp = dpt[r];
    if (DROP) {
      const int64_t kvc = mykv < seq ? mykv : seq - 1;
      const unsigned int w = maskrow[qgc * mask_w + (kvc >> 5)];
      dp *= ((w >> (kvc & 31)) & 1u) ? inv_keep : 0.f;
    }
    if (MASKED) {
      bool masked = qg >= seq || (causal && mykv > qg) || mykv >= seq;
      float pv = masked ? 0.f : __expf(st[r] * scale - lsep[qgc]);
      ds[r] = masked ? 0.f : pv * (dp - dltp[qgc]) * scale;
    } else {
      float pv = __expf(st[r] * scale - lsep[qg]);
      ds[r] = pv * (dp - dltp[qg]) * scale;
    }
  }
  bf16x8 db0 = assemble_pfrag(&ds[0]);
  bf16x8 db1 = assemble_pfrag(&ds[8]);
#pragma unroll
  for (int qc = 0; qc < 2; ++qc) {
    bf16x8 db = qc == 0 ? db0 : db1;
#pragma unroll
    for (int dj = 0; dj < D / 32; ++dj) {
      bf16x8 qt;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int qr = sub + qc * 16 + hi * 8 + j;
        qt[j] = ldsQ[qr][dj * 32 + lkv];
      }
      dkt[dj] =
          __builtin_amdgcn_mfma_f32_32x32x16_bf16(qt, db, dkt[dj], 0, 0, 0);
    }
  }
}

template <bool DROP, bool DBUF, int D>
__global__ __launch_bounds__(256, D == 64 ? (DBUF ? 3 : 4) : 2)
void attn_bwd_dv_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ dout, const float* __restrict__ lse,
    short* __restrict__ dv, int64_t seq, float scale, int causal,
    int64_t heads, int64_t in_sb, int64_t in_sh, int64_t in_ss,
    int64_t do_sb, int64_t do_sh, int64_t do_ss, int64_t g_sb,
    int64_t g_sh, int64_t g_ss,
    const unsigned int* __restrict__ mask, int64_t mask_w,
    float inv_keep) {
  __shared__ short ldsQ[DBUF ? 2 : 1][64][D + 8];
  __shared__ short ldsDO[DBUF ? 2 : 1][64][D + 8];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int lkv = lane & 31;
  // causal: kv0 rides the Y axis (ascending = longest q-range first),
  // bh on X — global longest-first packing like the fwd kernel
  const int64_t bh = causal ? (int64_t)blockIdx.x : (int64_t)blockIdx.y;
  const int64_t kv0_blk =
      (causal ? (int64_t)blockIdx.y : (int64_t)blockIdx.x) * 128;
  const int64_t kv0 = kv0_blk + wave * 32;
  const bool active = kv0 < seq;
  const int64_t boff = (bh / heads) * in_sb + (bh % heads) * in_sh;
  const short* qp = q + boff;
  const short* kp = k + boff;
  const short* dop = dout + (bh / heads) * do_sb + (bh % heads) * do_sh;
  const float* lsep = lse + bh * seq;

  const int64_t mykv = kv0 + lkv;
  const int64_t kvrow = mykv < seq ? mykv : seq - 1;
  bf16x8 kfrag[D / 16];
#pragma unroll
  for (int c = 0; c < D / 16; ++c)
    kfrag[c] = *reinterpret_cast<const bf16x8*>(
        kp + kvrow * in_ss + hi * 8 + 16 * c);

  f32x16 dvt[D / 32] = {};
  const int stage_row = threadIdx.x >> 3;
  const int stage_seg = (threadIdx.x & 7) * 8;

  // DBUF=false: stage 64 q rows straight to LDS per barrier pair.
  // DBUF=true: software pipeline — this chunk's rows were prefetched
  // into registers during the previous chunk's MFMAs; commit them to
  // the alternate LDS buffer, one barrier, then issue the next chunk's
  // global loads so they hide under this chunk's compute (the fwd v7
  // staging pattern applied to the q/dO loop).
  bf16x8 rq[2][D / 64], rdo[2][D / 64];
  auto prefetch = [&](int64_t q0) {
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      int64_t qr = q0 + stage_row + half * 32;
      if (qr >= seq) qr = seq - 1;
#pragma unroll
      for (int s = 0; s < D / 64; ++s) {
        rq[half][s] = *reinterpret_cast<const bf16x8*>(
            qp + qr * in_ss + stage_seg + s * 64);
        rdo[half][s] = *reinterpret_cast<const bf16x8*>(
            dop + qr * do_ss + stage_seg + s * 64);
      }
    }
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int half = 0; half < 2; ++half)
#pragma unroll
      for (int s = 0; s < D / 64; ++s) {
        *reinterpret_cast<bf16x8*>(
            &ldsQ[buf][stage_row + half * 32][stage_seg + s * 64]) =
            rq[half][s];
        *reinterpret_cast<bf16x8*>(
            &ldsDO[buf][stage_row + half * 32][stage_seg + s * 64]) =
            rdo[half][s];
      }
  };
  auto stage_q64 = [&](int64_t q0) {
    __syncthreads();
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      int64_t qr = q0 + stage_row + half * 32;
      if (qr >= seq) qr = seq - 1;
#pragma unroll
      for (int seg = 0; seg < D; seg += 64) {
        *reinterpret_cast<bf16x8*>(
            &ldsQ[0][stage_row + half * 32][stage_seg + seg]) =
            *reinterpret_cast<const bf16x8*>(
                qp + qr * in_ss + stage_seg + seg);
        *reinterpret_cast<bf16x8*>(
            &ldsDO[0][stage_row + half * 32][stage_seg + seg]) =
            *reinterpret_cast<const bf16x8*>(
                dop + qr * do_ss + stage_seg + seg);
      }
    }
    __syncthreads();
  };

  // unified 64-row chunk loop; the phase test (masked causal diagonal /
  // branch-free bulk / masked seq tail) is block-uniform per chunk
  const unsigned int* maskrow =
      DROP ? mask + bh * seq * mask_w : (const unsigned int*)nullptr;
  const int64_t diag_end =
      causal ? (kv0_blk + 128 < seq ? kv0_blk + 128 : seq) : 0;
  const int64_t bulk_end = seq & ~(int64_t)63;
  if (!DBUF) {
    // two-barrier direct staging, phase-split loops (the measured r2
    // baseline — kept verbatim so the fallback stays spill-free)
    int64_t q0 = causal ? kv0_blk : 0;
    for (; q0 < diag_end; q0 += 64) {
      stage_q64(q0);
      if (!active) continue;
#pragma unroll
      for (int sub = 0; sub < 64; sub += 32) {
        const int64_t q0s = q0 + sub;
        if (q0s >= diag_end || (causal && q0s + 31 < kv0)) continue;
        dv_tile<true, DROP, D>(ldsQ[0], ldsDO[0], sub, q0s, seq, mykv,
                               scale, causal, lsep, kfrag, dvt, lkv, hi,
                               maskrow, mask_w, inv_keep);
      }
    }
    q0 = diag_end > q0 ? diag_end : q0;
    for (; q0 + 63 < bulk_end; q0 += 64) {
      stage_q64(q0);
#pragma unroll
      for (int sub = 0; sub < 64; sub += 32)
        dv_tile<false, DROP, D>(ldsQ[0], ldsDO[0], sub, q0 + sub, seq,
                                mykv, scale, causal, lsep, kfrag, dvt,
                                lkv, hi, maskrow, mask_w, inv_keep);
    }
    for (; q0 < seq; q0 += 64) {
      stage_q64(q0);
      if (!active) continue;
#pragma unroll
      for (int sub = 0; sub < 64; sub += 32) {
        const int64_t q0s = q0 + sub;
        if (q0s >= seq) break;
        dv_tile<true, DROP, D>(ldsQ[0], ldsDO[0], sub, q0s, seq, mykv,
                               scale, causal, lsep, kfrag, dvt, lkv, hi,
                               maskrow, mask_w, inv_keep);
      }
    }
  } else {
    const int64_t q_begin = causal ? kv0_blk : 0;
    int buf = 0;
    prefetch(q_begin);
    for (int64_t q0 = q_begin; q0 < seq; q0 += 64) {
      commit(buf);
      __syncthreads();
      if (q0 + 64 < seq) prefetch(q0 + 64);
      const int rb = buf;
      buf ^= 1;
      const bool plain = q0 >= diag_end && q0 + 63 < bulk_end;
      if (plain) {
#pragma unroll
        for (int sub = 0; sub < 64; sub += 32)
          dv_tile<false, DROP, D>(ldsQ[rb], ldsDO[rb], sub, q0 + sub,
                                  seq, mykv, scale, causal, lsep, kfrag,
                                  dvt, lkv, hi, maskrow, mask_w,
                                  inv_keep);
      } else if (active) {
#pragma unroll
        for (int sub = 0; sub < 64; sub += 32) {
          const int64_t q0s = q0 + sub;
          if (q0s >= seq) break;
          if (causal && q0s + 31 < kv0) continue;
          dv_tile<true, DROP, D>(ldsQ[rb], ldsDO[rb], sub, q0s, seq,
                                 mykv, scale, causal, lsep, kfrag, dvt,
                                 lkv, hi, maskrow, mask_w, inv_keep);
        }
      }
    }
  }
  if (!active || mykv >= seq) return;
  short* dvp = dv + (bh / heads) * g_sb + (bh % heads) * g_sh +
               mykv * g_ss;
#pragma unroll
  for (int dj = 0; dj < D / 32; ++dj)
#pragma unroll
    for (int r = 0; r < 16; ++r)
      dvp[dj * 32 + crow(r, hi)] = (short)f2bf(dvt[dj][r]);
}

template <bool DROP, bool DBUF, int D>
__global__ __launch_bounds__(256, D == 64 ? 3 : 2) void attn_bwd_dk_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dk, int64_t seq, float scale, int causal,
    int64_t heads, int64_t in_sb, int64_t in_sh, int64_t in_ss,
    int64_t do_sb, int64_t do_sh, int64_t do_ss, int64_t g_sb,
    int64_t g_sh, int64_t g_ss,
    const unsigned int* __restrict__ mask, int64_t mask_w,
    float inv_keep) {
  __shared__ short ldsQ[DBUF ? 2 : 1][64][D + 8];
  __shared__ short ldsDO[DBUF ? 2 : 1][64][D + 8];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int lkv = lane & 31;
  const int64_t bh = causal ? (int64_t)blockIdx.x : (int64_t)blockIdx.y;
  const int64_t kv0_blk =
      (causal ? (int64_t)blockIdx.y : (int64_t)blockIdx.x) * 128;
  const int64_t kv0 = kv0_blk + wave * 32;
  const bool active = kv0 < seq;
  const int64_t boff = (bh / heads) * in_sb + (bh % heads) * in_sh;
  const short* qp = q + boff;
  const short* kp = k + boff;
  const short* vp = v + boff;
  const short* dop = dout + (bh / heads) * do_sb + (bh % heads) * do_sh;
  const float* lsep = lse + bh * seq;
  const float* dltp = delta + bh * seq;

  const int64_t mykv = kv0 + lkv;
  const int64_t kvrow = mykv < seq ? mykv : seq - 1;
  bf16x8 kfrag[D / 16], vfrag[D / 16];
#pragma unroll
  for (int c = 0; c < D / 16; ++c) {
    kfrag[c] = *reinterpret_cast<const bf16x8*>(
        kp + kvrow * in_ss + hi * 8 + 16 * c);
    vfrag[c] = *reinterpret_cast<const bf16x8*>(
        vp + kvrow * in_ss + hi * 8 + 16 * c);
  }

  f32x16 dkt[D / 32] = {};
  const int stage_row = threadIdx.x >> 3;
  const int stage_seg = (threadIdx.x & 7) * 8;

  // staging: same DBUF pipeline as the dV kernel (see comment there)
  bf16x8 rq[2][D / 64], rdo[2][D / 64];
  auto prefetch = [&](int64_t q0) {
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      int64_t qr = q0 + stage_row + half * 32;
      if (qr >= seq) qr = seq - 1;
#pragma unroll
      for (int s = 0; s < D / 64; ++s) {
        rq[half][s] = *reinterpret_cast<const bf16x8*>(
            qp + qr * in_ss + stage_seg + s * 64);
        rdo[half][s] = *reinterpret_cast<const bf16x8*>(
            dop + qr * do_ss + stage_seg + s * 64);
      }
    }
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int half = 0; half < 2; ++half)
#pragma unroll
      for (int s = 0; s < D / 64; ++s) {
        *reinterpret_cast<bf16x8*>(
            &ldsQ[buf][stage_row + half * 32][stage_seg + s * 64]) =
            rq[half][s];
        *reinterpret_cast<bf16x8*>(
            &ldsDO[buf][stage_row + half * 32][stage_seg + s * 64]) =
            rdo[half][s];
      }
  };
  auto stage_q64 = [&](int64_t q0) {
    __syncthreads();
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      int64_t qr = q0 + stage_row + half * 32;
      if (qr >= seq) qr = seq - 1;
#pragma unroll
      for (int seg = 0; seg < D; seg += 64) {
        *reinterpret_cast<bf16x8*>(
            &ldsQ[0][stage_row + half * 32][stage_seg + seg]) =
            *reinterpret_cast<const bf16x8*>(
                qp + qr * in_ss + stage_seg + seg);
        *reinterpret_cast<bf16x8*>(
            &ldsDO[0][stage_row + half * 32][stage_seg + seg]) =
            *reinterpret_cast<const bf16x8*>(
                dop + qr * do_ss + stage_seg + seg);
      }
    }
    __syncthreads();
  };

  const unsigned int* maskrow =
      DROP ? mask + bh * seq * mask_w : (const unsigned int*)nullptr;
  const int64_t diag_end =
      causal ? (kv0_blk + 128 < seq ? kv0_blk + 128 : seq) : 0;
  const int64_t bulk_end = seq & ~(int64_t)63;
  if (!DBUF) {
    int64_t q0 = causal ? kv0_blk : 0;
    for (; q0 < diag_end; q0 += 64) {
      stage_q64(q0);
      if (!active) continue;
#pragma unroll
      for (int sub = 0; sub < 64; sub += 32) {
        const int64_t q0s = q0 + sub;
        if (q0s >= diag_end || (causal && q0s + 31 < kv0)) continue;
        dk_tile<true, DROP, D>(ldsQ[0], ldsDO[0], sub, q0s, seq, mykv,
                               scale, causal, lsep, dltp, kfrag, vfrag,
                               dkt, lkv, hi, maskrow, mask_w, inv_keep);
      }
    }
    q0 = diag_end > q0 ? diag_end : q0;
    for (; q0 + 63 < bulk_end; q0 += 64) {
      stage_q64(q0);
#pragma unroll
      for (int sub = 0; sub < 64; sub += 32)
        dk_tile<false, DROP, D>(ldsQ[0], ldsDO[0], sub, q0 + sub, seq,
                                mykv, scale, causal, lsep, dltp, kfrag,
                                vfrag, dkt, lkv, hi, maskrow, mask_w,
                                inv_keep);
    }
    for (; q0 < seq; q0 += 64) {
      stage_q64(q0);
      if (!active) continue;
#pragma unroll
      for (int sub = 0; sub < 64; sub += 32) {
        const int64_t q0s = q0 + sub;
        if (q0s >= seq) break;
        dk_tile<true, DROP, D>(ldsQ[0], ldsDO[0], sub, q0s, seq, mykv,
                               scale, causal, lsep, dltp, kfrag, vfrag,
                               dkt, lkv, hi, maskrow, mask_w, inv_keep);
      }
    }
  } else {
    const int64_t q_begin = causal ? kv0_blk : 0;
    int buf = 0;
    prefetch(q_begin);
    for (int64_t q0 = q_begin; q0 < seq; q0 += 64) {
      commit(buf);
      __syncthreads();
      if (q0 + 64 < seq) prefetch(q0 + 64);
      const int rb = buf;
      buf ^= 1;
      const bool plain = q0 >= diag_end && q0 + 63 < bulk_end;
      if (plain) {
#pragma unroll
        for (int sub = 0; sub < 64; sub += 32)
          dk_tile<false, DROP, D>(ldsQ[rb], ldsDO[rb], sub, q0 + sub,
                                  seq, mykv, scale, causal, lsep, dltp,
                                  kfrag, vfrag, dkt, lkv, hi, maskrow,
                                  mask_w, inv_keep);
      } else if (active) {
#pragma unroll
        for (int sub = 0; sub < 64; sub += 32) {
          const int64_t q0s = q0 + sub;
          if (q0s >= seq) break;
          if (causal && q0s + 31 < kv0) continue;
          dk_tile<true, DROP, D>(ldsQ[rb], ldsDO[rb], sub, q0s, seq,
                                 mykv, scale, causal, lsep, dltp, kfrag,
                                 vfrag, dkt, lkv, hi, maskrow, mask_w,
                                 inv_keep);
        }
      }
    }
  }
  if (!active || mykv >= seq) return;
  short* dkp = dk + (bh / heads) * g_sb + (bh % heads) * g_sh +
               mykv * g_ss;
#pragma unroll
  for (int dj = 0; dj < D / 32; ++dj)
#pragma unroll
    for (int r = 0; r < 16; ++r)
      dkp[dj * 32 + crow(r, hi)] = (short)f2bf(dkt[dj][r]);
}

// ============================================================================
// backward kernel B (q-parallel): dQ.  K and V tiles staged in LDS as
// row copies per block; row fragments via ds_read_b128, K-transposed
// fragments via 2-byte LDS reads.  Like the forward, the kv loop is
// split into a branch-free bulk phase (below the causal diagonal /
// inside the seq bound) and a masked diagonal+tail phase.
// ============================================================================

// one 32-key sub-tile (of the 64-row staged chunk) of the dQ loop.
// DROP: dS = P o (dP_d o M/(1-p) - D) — P stays undropped, the V^T dO
// term is masked+rescaled with the forward's published keep bits.
template <bool MASKED, bool DROP, int D>
DEVI void dq_tile(const short (&ldsK)[64][D + 8],
                  const short (&ldsVr)[64][D + 8],
                  int sub, int64_t kv0, int64_t seq, int64_t myq,
                  float scale, int causal, float mylse, float mydelta,
                  const bf16x8 (&qfrag)[D / 16],
                  const bf16x8 (&dofrag)[D / 16],
                  f32x16 (&dqt)[D / 32], int lq, int hi,
                  const unsigned int* maskrow, int64_t mask_w,
                  float inv_keep) {
  f32x16 st = {};
  f32x16 dpt = {};
#pragma unroll
  for (int c = 0; c < D / 16; ++c) {
    bf16x8 kf = *reinterpret_cast<const bf16x8*>(
        &ldsK[sub + lq][hi * 8 + 16 * c]);
    bf16x8 vf = *reinterpret_cast<const bf16x8*>(
        &ldsVr[sub + lq][hi * 8 + 16 * c]);
    st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qfrag[c], st, 0, 0, 0);
    dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dofrag[c], dpt, 0, 0,
                                                  0);
  }
  unsigned int mword = 0;
  if (DROP) {
    const int64_t qc = myq < seq ? myq : seq - 1;
    mword = maskrow[qc * mask_w + (kv0 >> 5)];
  }
  float ds[16];
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    float pv = __expf(st[r] * scale - mylse);
    if (MASKED) {
      const int64_t kvg = kv0 + crow(r, hi);
      if (myq >= seq || kvg >= seq || (causal && kvg > myq)) pv = 0.f;
    }
    float dp = dpt[r];
    if (DROP)
      dp *= ((mword >> crow(r, hi)) & 1u) ? inv_keep : 0.f;
    ds[r] = pv * (dp - mydelta) * scale;
  }
  bf16x8 db0 = assemble_pfrag(&ds[0]);
  bf16x8 db1 = assemble_pfrag(&ds[8]);
#pragma unroll
  for (int kc = 0; kc < 2; ++kc) {
    bf16x8 db = kc == 0 ? db0 : db1;
#pragma unroll
    for (int dj = 0; dj < D / 32; ++dj) {
      bf16x8 kt;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int kr = sub + kc * 16 + hi * 8 + j;
        kt[j] = ldsK[kr][dj * 32 + lq];
      }
      dqt[dj] =
          __builtin_amdgcn_mfma_f32_32x32x16_bf16(kt, db, dqt[dj], 0, 0, 0);
    }
  }
}

template <bool DROP, bool DBUF, int D>
__global__ __launch_bounds__(256, D == 64 ? 3 : 2) void attn_bwd_dq_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dq, const short* __restrict__ out, int64_t seq,
    float scale, int causal, int64_t heads, int64_t in_sb, int64_t in_sh,
    int64_t in_ss, int64_t do_sb, int64_t do_sh, int64_t do_ss,
    int64_t g_sb, int64_t g_sh, int64_t g_ss, int64_t o_sb, int64_t o_sh,
    int64_t o_ss, const unsigned int* __restrict__ mask, int64_t mask_w,
    float inv_keep, const float* __restrict__ dlse) {
  __shared__ short ldsK[DBUF ? 2 : 1][64][D + 8];
  __shared__ short ldsVr[DBUF ? 2 : 1][64][D + 8];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;
  // causal: global longest-first packing (see attn_fwd_kernel)
  const int64_t bh = causal ? (int64_t)blockIdx.x : (int64_t)blockIdx.y;
  const int64_t qb = causal ? (int64_t)gridDim.y - 1 - blockIdx.y
                            : (int64_t)blockIdx.x;
  const int64_t q0_blk = qb * 128;
  const int64_t q0 = q0_blk + wave * 32;
  const bool active = q0 < seq;
  const int64_t boff = (bh / heads) * in_sb + (bh % heads) * in_sh;
  const short* qp = q + boff;
  const short* kp = k + boff;
  const short* vp = v + boff;
  const short* dop = dout + (bh / heads) * do_sb + (bh % heads) * do_sh;

  const int64_t myq = q0 + lq;
  const int64_t qrow = myq < seq ? myq : seq - 1;
  const float mylse = lse[bh * seq + qrow];

  const short* op_ = out + (bh / heads) * o_sb + (bh % heads) * o_sh;
  bf16x8 qfrag[D / 16], dofrag[D / 16];
  float dsum = 0.f;
#pragma unroll
  for (int c = 0; c < D / 16; ++c) {
    qfrag[c] = *reinterpret_cast<const bf16x8*>(
        qp + qrow * in_ss + hi * 8 + 16 * c);
    dofrag[c] = *reinterpret_cast<const bf16x8*>(
        dop + qrow * do_ss + hi * 8 + 16 * c);
    bf16x8 of = *reinterpret_cast<const bf16x8*>(
        op_ + qrow * o_ss + hi * 8 + 16 * c);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      dsum += bf2f(dofrag[c][j]) * bf2f(of[j]);
  }
  // D_i = rowsum(dO * O): this lane covers 32 of the 64 d-elements, the
  // partner lane (^32) the rest
  // dlse (ring-attention block merge): lse = logsumexp(S) has
  // dlse_i/dS_ij = P_ij, so an incoming lse gradient folds into the
  // per-row constant: dS = P o (dP - (D - dlse)).  The published delta
  // is pre-adjusted so the dK kernel needs no change.
  float mydelta = dsum + __shfl_xor(dsum, 32, 64);
  if (dlse != nullptr) mydelta -= dlse[bh * seq + qrow];
  // publish delta for the dK kernel that follows on the same stream
  if (myq < seq && hi == 0)
    const_cast<float*>(delta)[bh * seq + myq] = mydelta;

  f32x16 dqt[D / 32] = {};

  const int stage_row = threadIdx.x >> 3;
  const int stage_seg = (threadIdx.x & 7) * 8;
  // bulk chunks (64 kv rows) are full for every lane; q0_blk is a
  // multiple of 128 so the causal bulk region is 64-aligned
  const int64_t bulk_end = causal ? q0_blk : (seq & ~(int64_t)63);
  const int64_t blk_kv_end =
      causal ? (q0_blk + 128 < seq ? q0_blk + 128 : seq) : seq;

  auto stage_kv64 = [&](int64_t kv0) {
    __syncthreads();
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      int64_t kr = kv0 + stage_row + half * 32;
      if (kr >= seq) kr = seq - 1;   // dS there is 0
#pragma unroll
      for (int seg = 0; seg < D; seg += 64) {
        *reinterpret_cast<bf16x8*>(
            &ldsK[0][stage_row + half * 32][stage_seg + seg]) =
            *reinterpret_cast<const bf16x8*>(
                kp + kr * in_ss + stage_seg + seg);
        *reinterpret_cast<bf16x8*>(
            &ldsVr[0][stage_row + half * 32][stage_seg + seg]) =
            *reinterpret_cast<const bf16x8*>(
                vp + kr * in_ss + stage_seg + seg);
      }
    }
    __syncthreads();
  };

  // DBUF: same register-double-buffered single-barrier pipeline as the
  // dV/dK kernels (next kv chunk prefetched into VGPRs during this
  // chunk's MFMAs; see attn_bwd_dv_kernel).
  bf16x8 rk[2][D / 64], rv[2][D / 64];
  auto prefetch = [&](int64_t kv0) {
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      int64_t kr = kv0 + stage_row + half * 32;
      if (kr >= seq) kr = seq - 1;
#pragma unroll
      for (int s = 0; s < D / 64; ++s) {
        rk[half][s] = *reinterpret_cast<const bf16x8*>(
            kp + kr * in_ss + stage_seg + s * 64);
        rv[half][s] = *reinterpret_cast<const bf16x8*>(
            vp + kr * in_ss + stage_seg + s * 64);
      }
    }
  };
  auto commit = [&](int buf) {
#pragma unroll
    for (int half = 0; half < 2; ++half)
#pragma unroll
      for (int s = 0; s < D / 64; ++s) {
        *reinterpret_cast<bf16x8*>(
            &ldsK[buf][stage_row + half * 32][stage_seg + s * 64]) =
            rk[half][s];
        *reinterpret_cast<bf16x8*>(
            &ldsVr[buf][stage_row + half * 32][stage_seg + s * 64]) =
            rv[half][s];
      }
  };

  const unsigned int* maskrow =
      DROP ? mask + bh * seq * mask_w : (const unsigned int*)nullptr;
  if (!DBUF) {
    int64_t kv0 = 0;
    for (; kv0 < bulk_end; kv0 += 64) {
      stage_kv64(kv0);
#pragma unroll
      for (int sub = 0; sub < 64; sub += 32)
        dq_tile<false, DROP, D>(ldsK[0], ldsVr[0], sub, kv0 + sub, seq,
                                myq, scale, causal, mylse, mydelta,
                                qfrag, dofrag, dqt, lq, hi, maskrow,
                                mask_w, inv_keep);
    }
    for (; kv0 < blk_kv_end; kv0 += 64) {
      stage_kv64(kv0);
      const int64_t wave_kv_end = causal
          ? (q0 + 32 < seq ? q0 + 32 : seq) : seq;
      if (!active) continue;
#pragma unroll
      for (int sub = 0; sub < 64; sub += 32) {
        const int64_t kvs = kv0 + sub;
        if (kvs >= wave_kv_end) break;
        dq_tile<true, DROP, D>(ldsK[0], ldsVr[0], sub, kvs, seq, myq,
                               scale, causal, mylse, mydelta, qfrag,
                               dofrag, dqt, lq, hi, maskrow, mask_w,
                               inv_keep);
      }
    }
  } else {
    int buf = 0;
    prefetch(0);
    for (int64_t kv0 = 0; kv0 < blk_kv_end; kv0 += 64) {
      commit(buf);
      __syncthreads();
      if (kv0 + 64 < blk_kv_end) prefetch(kv0 + 64);
      const int rb = buf;
      buf ^= 1;
      if (kv0 < bulk_end) {
#pragma unroll
        for (int sub = 0; sub < 64; sub += 32)
          dq_tile<false, DROP, D>(ldsK[rb], ldsVr[rb], sub, kv0 + sub,
                                  seq, myq, scale, causal, mylse,
                                  mydelta, qfrag, dofrag, dqt, lq, hi,
                                  maskrow, mask_w, inv_keep);
      } else if (active) {
        const int64_t wave_kv_end = causal
            ? (q0 + 32 < seq ? q0 + 32 : seq) : seq;
#pragma unroll
        for (int sub = 0; sub < 64; sub += 32) {
          const int64_t kvs = kv0 + sub;
          if (kvs >= wave_kv_end) break;
          dq_tile<true, DROP, D>(ldsK[rb], ldsVr[rb], sub, kvs, seq,
                                 myq, scale, causal, mylse, mydelta,
                                 qfrag, dofrag, dqt, lq, hi, maskrow,
                                 mask_w, inv_keep);
        }
      }
    }
  }

  if (!active || myq >= seq) return;
  short* dqp = dq + (bh / heads) * g_sb + (bh % heads) * g_sh + myq * g_ss;
#pragma unroll
  for (int dj = 0; dj < D / 32; ++dj)
#pragma unroll
    for (int r = 0; r < 16; ++r)
      dqp[dj * 32 + crow(r, hi)] = (short)f2bf(dqt[dj][r]);
}

}  // namespace

extern "C" {

void epl_attn_fwd(const void* q, const void* k, const void* v, void* out,
                  float* lse, int64_t batch_heads, int64_t seq, float scale,
                  bool causal, int64_t heads, int64_t head_dim,
                  const int64_t* in_strides,
                  const int64_t* o_strides, unsigned int* drop_mask,
                  int64_t mask_w, unsigned long long seed, int drop_thresh,
                  float inv_keep, hipStream_t stream) {
  // causal: (x=bh, y=qblocks) — x-fastest dispatch launches every
  // batch-head of the longest q-block first (kernels read the mapping
  // from the causal flag)
  const unsigned nqb = (unsigned)((seq + 127) / 128);
  dim3 grid = causal ? dim3((unsigned)batch_heads, nqb)
                     : dim3(nqb, (unsigned)batch_heads);
#define FWD_ARGS                                                         \
  reinterpret_cast<const short*>(q), reinterpret_cast<const short*>(k),  \
      reinterpret_cast<const short*>(v), reinterpret_cast<short*>(out),  \
      lse, seq, scale, causal ? 1 : 0, heads, in_strides[0],             \
      in_strides[1], in_strides[2], o_strides[0], o_strides[1],          \
      o_strides[2], drop_mask, mask_w, seed, drop_thresh, inv_keep
  if (head_dim == 128) {
    if (drop_mask != nullptr)
      hipLaunchKernelGGL(HIP_KERNEL_NAME(attn_fwd_kernel<true, 128>),
                         grid, dim3(256), 0, stream, FWD_ARGS);
    else
      hipLaunchKernelGGL(HIP_KERNEL_NAME(attn_fwd_kernel<false, 128>),
                         grid, dim3(256), 0, stream, FWD_ARGS);
  } else if (drop_mask != nullptr) {
    hipLaunchKernelGGL(HIP_KERNEL_NAME(attn_fwd_kernel<true, 64>), grid,
                       dim3(256), 0, stream, FWD_ARGS);
  } else {
    hipLaunchKernelGGL(HIP_KERNEL_NAME(attn_fwd_kernel<false, 64>), grid,
                       dim3(256), 0, stream, FWD_ARGS);
  }
#undef FWD_ARGS
}

void epl_attn_bwd(const void* q, const void* k, const void* v,
                  const void* out, const void* dout, const float* lse,
                  float* delta_ws, void* dq, void* dk, void* dv,
                  int64_t batch_heads, int64_t seq, float scale,
                  bool causal, int64_t heads, const int64_t* in_strides,
                  const int64_t* o_strides, const int64_t* do_strides,
                  const int64_t* g_strides, int split_dkdv,
                  const unsigned int* drop_mask, int64_t mask_w,
                  float inv_keep, int64_t head_dim, const float* dlse,
                  hipStream_t stream) {
  const int64_t rows = batch_heads * seq;
  const unsigned nqb = (unsigned)((seq + 127) / 128);
  dim3 grid = causal ? dim3((unsigned)batch_heads, nqb)
                     : dim3(nqb, (unsigned)batch_heads);
  // head_dim 128 and causal always run the split kernels (the combined
  // dkdv kernel is d=64-only and keeps the legacy x-major grid)
  if (split_dkdv || drop_mask != nullptr || head_dim == 128 || causal ||
      dlse != nullptr) {
    // order: dV (needs no delta) -> dQ (computes + publishes delta from
    // the dO/O rows it already loads) -> dK (consumes delta).  The prep
    // pass disappears.  Dropout always runs the split kernels (the
    // combined dkdv kernel has no mask path).
#define DV_ARGS                                                          \
  reinterpret_cast<const short*>(q), reinterpret_cast<const short*>(k),  \
      reinterpret_cast<const short*>(dout), lse,                         \
      reinterpret_cast<short*>(dv), seq, scale, causal ? 1 : 0, heads,   \
      in_strides[0], in_strides[1], in_strides[2], do_strides[0],        \
      do_strides[1], do_strides[2], g_strides[0], g_strides[1],          \
      g_strides[2], drop_mask, mask_w, inv_keep
#define DQ_ARGS                                                          \
  reinterpret_cast<const short*>(q), reinterpret_cast<const short*>(k),  \
      reinterpret_cast<const short*>(v),                                 \
      reinterpret_cast<const short*>(dout), lse, delta_ws,               \
      reinterpret_cast<short*>(dq), reinterpret_cast<const short*>(out), \
      seq, scale, causal ? 1 : 0, heads, in_strides[0], in_strides[1],   \
      in_strides[2], do_strides[0], do_strides[1], do_strides[2],        \
      g_strides[0], g_strides[1], g_strides[2], o_strides[0],            \
      o_strides[1], o_strides[2], drop_mask, mask_w, inv_keep, dlse
#define DK_ARGS                                                          \
  reinterpret_cast<const short*>(q), reinterpret_cast<const short*>(k),  \
      reinterpret_cast<const short*>(v),                                 \
      reinterpret_cast<const short*>(dout), lse, delta_ws,               \
      reinterpret_cast<short*>(dk), seq, scale, causal ? 1 : 0, heads,   \
      in_strides[0], in_strides[1], in_strides[2], do_strides[0],        \
      do_strides[1], do_strides[2], g_strides[0], g_strides[1],          \
      g_strides[2], drop_mask, mask_w, inv_keep
    // d64 staging mode (EPL_ATTN_BWD_DBUF): 0 = two-barrier direct
    // staging; 1 = dV double-buffered single-barrier pipeline
    // (spill-free); 2 = dK pipelined too (spills 12-40 VGPRs at 3
    // waves but still wins on long q loops); 3 = dQ pipelined too
    // (spills 68 B/lane at 3 waves).
    // Default -1 = auto: non-causal runs mode 2 (same-box A/B in
    // profiles/r02_dbuf_ab.txt: bert s512 -7.6%, s4096 -9.9%), causal
    // keeps direct staging (its per-block q loop is half as long and
    // mode 2 measured +2% there).  d128 keeps direct staging (doubled
    // LDS would cost its 2-waves occupancy).
    static const int dbuf_mode = [] {
      const char* e = getenv("EPL_ATTN_BWD_DBUF");
      return e == nullptr ? -1 : atoi(e);
    }();
    const int mode = dbuf_mode < 0 ? (causal ? 0 : 2) : dbuf_mode;
    const bool dbuf = mode >= 1;
    const bool dbuf_k = mode >= 2;
    const bool dbuf_q = mode >= 3;
    if (head_dim == 128) {
      if (drop_mask != nullptr) {
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dv_kernel<true, false, 128>), grid,
            dim3(256), 0, stream, DV_ARGS);
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dq_kernel<true, false, 128>), grid,
            dim3(256), 0, stream, DQ_ARGS);
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dk_kernel<true, false, 128>), grid,
            dim3(256), 0, stream, DK_ARGS);
      } else {
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dv_kernel<false, false, 128>), grid,
            dim3(256), 0, stream, DV_ARGS);
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dq_kernel<false, false, 128>), grid,
            dim3(256), 0, stream, DQ_ARGS);
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dk_kernel<false, false, 128>), grid,
            dim3(256), 0, stream, DK_ARGS);
      }
    } else if (drop_mask != nullptr) {
      if (dbuf)
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dv_kernel<true, true, 64>), grid,
            dim3(256), 0, stream, DV_ARGS);
      else
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dv_kernel<true, false, 64>), grid,
            dim3(256), 0, stream, DV_ARGS);
      if (dbuf_q)
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dq_kernel<true, true, 64>), grid,
            dim3(256), 0, stream, DQ_ARGS);
      else
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dq_kernel<true, false, 64>), grid,
            dim3(256), 0, stream, DQ_ARGS);
      if (dbuf_k)
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dk_kernel<true, true, 64>), grid,
            dim3(256), 0, stream, DK_ARGS);
      else
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dk_kernel<true, false, 64>), grid,
            dim3(256), 0, stream, DK_ARGS);
    } else {
      if (dbuf)
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dv_kernel<false, true, 64>), grid,
            dim3(256), 0, stream, DV_ARGS);
      else
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dv_kernel<false, false, 64>), grid,
            dim3(256), 0, stream, DV_ARGS);
      if (dbuf_q)
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dq_kernel<false, true, 64>), grid,
            dim3(256), 0, stream, DQ_ARGS);
      else
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dq_kernel<false, false, 64>), grid,
            dim3(256), 0, stream, DQ_ARGS);
      if (dbuf_k)
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dk_kernel<false, true, 64>), grid,
            dim3(256), 0, stream, DK_ARGS);
      else
        hipLaunchKernelGGL(
            HIP_KERNEL_NAME(attn_bwd_dk_kernel<false, false, 64>), grid,
            dim3(256), 0, stream, DK_ARGS);
    }
#undef DV_ARGS
#undef DQ_ARGS
#undef DK_ARGS
    return;
  }
  {
    const int64_t blocks = (rows + 3) / 4;
    hipLaunchKernelGGL(attn_bwd_prep_kernel,
                       dim3((unsigned)(blocks < 4096 ? blocks : 4096)),
                       dim3(256), 0, stream,
                       reinterpret_cast<const short*>(dout),
                       reinterpret_cast<const short*>(out), delta_ws, rows,
                       seq, heads, do_strides[0], do_strides[1],
                       do_strides[2], o_strides[0], o_strides[1],
                       o_strides[2]);
  }
  hipLaunchKernelGGL(attn_bwd_dkdv_kernel, grid, dim3(256), 0, stream,
                     reinterpret_cast<const short*>(q),
                     reinterpret_cast<const short*>(k),
                     reinterpret_cast<const short*>(v),
                     reinterpret_cast<const short*>(dout), lse, delta_ws,
                     reinterpret_cast<short*>(dk),
                     reinterpret_cast<short*>(dv), seq, scale,
                     causal ? 1 : 0, heads, in_strides[0], in_strides[1],
                     in_strides[2], do_strides[0], do_strides[1],
                     do_strides[2], g_strides[0], g_strides[1],
                     g_strides[2]);
  hipLaunchKernelGGL(HIP_KERNEL_NAME(attn_bwd_dq_kernel<false, false, 64>),
                     grid, dim3(256), 0, stream,
                     reinterpret_cast<const short*>(q),
                     reinterpret_cast<const short*>(k),
                     reinterpret_cast<const short*>(v),
                     reinterpret_cast<const short*>(dout), lse, delta_ws,
                     reinterpret_cast<short*>(dq),
                     reinterpret_cast<const short*>(out), seq, scale,
                     causal ? 1 : 0, heads, in_strides[0], in_strides[1],
                     in_strides[2], do_strides[0], do_strides[1],
                     do_strides[2], g_strides[0], g_strides[1],
                     g_strides[2], o_strides[0], o_strides[1],
                     o_strides[2], drop_mask, mask_w, inv_keep, dlse);
}

}  // extern "C"
