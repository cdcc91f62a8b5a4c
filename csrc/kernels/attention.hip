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
// Backward: one kernel parallel over kv tiles; per q tile it recomputes
// P from Q,K,LSE, accumulates dV += P^T dO and dK += dS^T Q locally, and
// scatters dQ += dS K with fp32 global atomics (dq workspace); host casts
// dq to bf16 afterwards.  D_i = rowsum(dO*O) is precomputed by a small
// kernel.

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

// ============================================================================
// forward
// ============================================================================
__global__ __launch_bounds__(256) void attn_fwd_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, short* __restrict__ out,
    float* __restrict__ lse, int64_t seq, float scale, int causal) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;            // my q row within the tile
  const int64_t bh = blockIdx.y;
  const int64_t q0 = (int64_t)blockIdx.x * 128 + wave * 32;
  if (q0 >= seq) return;
  const short* qp = q + (bh * seq) * 64;
  const short* kp = k + (bh * seq) * 64;
  const short* vp = v + (bh * seq) * 64;

  const int64_t myq = q0 + lq;         // my global q row (may be >= seq)
  const int64_t qrow = myq < seq ? myq : seq - 1;

  // Q^T operand: B[k=d][n=q]; lane holds Q[qrow][hi*8+j + 16*c], c=0..3
  bf16x8 qfrag[4];
#pragma unroll
  for (int c = 0; c < 4; ++c)
    qfrag[c] = *reinterpret_cast<const bf16x8*>(
        qp + qrow * 64 + hi * 8 + 16 * c);

  f32x16 ot0 = {}, ot1 = {};           // O^T accumulators, d 0..31 / 32..63
  float m = -1e30f, l = 0.f;

  const int64_t kv_end = causal ? (q0 + 32 < seq ? q0 + 32 : seq)
                                : seq;
  for (int64_t kv0 = 0; kv0 < kv_end; kv0 += 32) {
    // ---- S^T = K @ Q^T ----------------------------------------------------
    f32x16 st = {};
    const int64_t krow = kv0 + lq < seq ? kv0 + lq : seq - 1;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      bf16x8 kfrag = *reinterpret_cast<const bf16x8*>(
          kp + krow * 64 + hi * 8 + 16 * c);
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfrag, qfrag[c], st, 0,
                                                   0, 0);
    }
    // ---- online softmax (lane owns q row; partner lane^32 has the other
    // 16 kv). st[reg] = S[myq][kv0 + crow(reg, hi)] ------------------------
    float s[16];
    float tile_max = -1e30f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      float sv = st[r] * scale;
      const int64_t kvg = kv0 + crow(r, hi);
      if (kvg >= seq || (causal && kvg > myq)) sv = -1e30f;
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
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      ot0[r] *= alpha;
      ot1[r] *= alpha;
    }
    // ---- O^T += V^T @ P^T -------------------------------------------------
    // two kv chunks of 16; P frag from s[0..7] / s[8..15]
    bf16x8 pf0 = assemble_pfrag(&s[0]);
    bf16x8 pf1 = assemble_pfrag(&s[8]);
    // A = V^T[d][kv]: lane holds V[kv0 + kc*16 + hi*8 + j][dchunk*32 + lq]
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      bf16x8 vt0, vt1;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int64_t kvg = kv0 + kc * 16 + hi * 8 + j;
        if (kvg >= seq) kvg = seq - 1;  // P there is 0
        vt0[j] = vp[kvg * 64 + lq];
        vt1[j] = vp[kvg * 64 + 32 + lq];
      }
      bf16x8 pf = kc == 0 ? pf0 : pf1;
      ot0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vt0, pf, ot0, 0, 0, 0);
      ot1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vt1, pf, ot1, 0, 0, 0);
    }
  }

  if (myq >= seq) return;
  const float inv_l = 1.f / l;
  short* op = out + (bh * seq + myq) * 64;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    op[crow(r, hi)] = (short)f2bf(ot0[r] * inv_l);
    op[32 + crow(r, hi)] = (short)f2bf(ot1[r] * inv_l);
  }
  if (hi == 0) lse[bh * seq + myq] = m + __logf(l);
}

// ============================================================================
// backward: D_i = rowsum(dO * O) precompute
// ============================================================================
__global__ void attn_bwd_prep_kernel(const short* __restrict__ dout,
                                     const short* __restrict__ out,
                                     float* __restrict__ delta,
                                     int64_t rows) {
  const int lane = threadIdx.x & 63;
  const int waves_per_block = blockDim.x >> 6;
  for (int64_t row = (int64_t)blockIdx.x * waves_per_block +
                     (threadIdx.x >> 6);
       row < rows; row += (int64_t)gridDim.x * waves_per_block) {
    const short* dp = dout + row * 64;
    const short* op = out + row * 64;
    float acc = bf2f(dp[lane]) * bf2f(op[lane]);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      acc += __shfl_down(acc, off, 64);
    if (lane == 0) delta[row] = acc;
  }
}

// ============================================================================
// backward kernel A (kv-parallel): dK and dV.  One wave per 32-key kv
// tile, looping q tiles.  Lane owns one kv row (lane%32), so dK/dV
// epilogues are direct stores.
//   P = exp(S*scale - lse);  dP = dO V^T;  dS = P*(dP - D_i)*scale
//   dV^T += dO^T @ P ;  dK^T += Q^T @ dS
// ============================================================================
__global__ __launch_bounds__(256) void attn_bwd_dkdv_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dk, short* __restrict__ dv, int64_t seq,
    float scale, int causal) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int lkv = lane & 31;           // my kv row within the tile
  const int64_t bh = blockIdx.y;
  const int64_t kv0 = (int64_t)blockIdx.x * 128 + wave * 32;
  if (kv0 >= seq) return;
  const short* qp = q + (bh * seq) * 64;
  const short* kp = k + (bh * seq) * 64;
  const short* vp = v + (bh * seq) * 64;
  const short* dop = dout + (bh * seq) * 64;
  const float* lsep = lse + bh * seq;
  const float* dltp = delta + bh * seq;

  const int64_t mykv = kv0 + lkv;
  const int64_t kvrow = mykv < seq ? mykv : seq - 1;

  // B operands for S^T/dP^T-style matmuls: lane holds row kvrow segments
  bf16x8 kfrag[4], vfrag[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    kfrag[c] = *reinterpret_cast<const bf16x8*>(
        kp + kvrow * 64 + hi * 8 + 16 * c);
    vfrag[c] = *reinterpret_cast<const bf16x8*>(
        vp + kvrow * 64 + hi * 8 + 16 * c);
  }

  f32x16 dvt0 = {}, dvt1 = {};   // dV^T acc: rows d 0..31 / 32..63, col kv
  f32x16 dkt0 = {}, dkt1 = {};   // dK^T acc

  const int64_t q_start = causal ? (kv0 / 32) * 32 : 0;
  for (int64_t q0 = q_start; q0 < seq; q0 += 32) {
    const int64_t qrow = q0 + lkv < seq ? q0 + lkv : seq - 1;
    // S[q][kv]: A = Q[m=q][k=d] (lane: q = q0+lane%32, contiguous d seg),
    // B = K^T[k=d][n=kv] (kfrag).  D: rows q = crow(r,hi), col kv = lkv.
    f32x16 st = {};
    f32x16 dpt = {};
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      bf16x8 qf = *reinterpret_cast<const bf16x8*>(
          qp + qrow * 64 + hi * 8 + 16 * c);
      bf16x8 dof = *reinterpret_cast<const bf16x8*>(
          dop + qrow * 64 + hi * 8 + 16 * c);
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf, kfrag[c], st, 0, 0,
                                                   0);
      dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dof, vfrag[c], dpt, 0,
                                                    0, 0);
    }
    // st[r] = S[q0+crow(r,hi)][mykv]; dpt[r] = dP same layout
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
    // B operands over q: element j = q row (qc*16 + hi*8 + j)
    bf16x8 pb0 = assemble_pfrag(&p[0]);
    bf16x8 pb1 = assemble_pfrag(&p[8]);
    bf16x8 db0 = assemble_pfrag(&ds[0]);
    bf16x8 db1 = assemble_pfrag(&ds[8]);
#pragma unroll
    for (int qc = 0; qc < 2; ++qc) {
      bf16x8 dot0, dot1, qt0, qt1;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int64_t qg = q0 + qc * 16 + hi * 8 + j;
        if (qg >= seq) qg = seq - 1;   // P/dS there is 0
        dot0[j] = dop[qg * 64 + lkv];
        dot1[j] = dop[qg * 64 + 32 + lkv];
        qt0[j] = qp[qg * 64 + lkv];
        qt1[j] = qp[qg * 64 + 32 + lkv];
      }
      bf16x8 pb = qc == 0 ? pb0 : pb1;
      bf16x8 db = qc == 0 ? db0 : db1;
      // dV^T[d][kv] += dO^T[d][q] @ P[q][kv]
      dvt0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dot0, pb, dvt0, 0, 0,
                                                     0);
      dvt1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dot1, pb, dvt1, 0, 0,
                                                     0);
      // dK^T[d][kv] += Q^T[d][q] @ dS[q][kv]
      dkt0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qt0, db, dkt0, 0, 0,
                                                     0);
      dkt1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qt1, db, dkt1, 0, 0,
                                                     0);
    }
  }

  if (mykv >= seq) return;
  short* dkp = dk + (bh * seq + mykv) * 64;
  short* dvp = dv + (bh * seq + mykv) * 64;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    dvp[crow(r, hi)] = (short)f2bf(dvt0[r]);
    dvp[32 + crow(r, hi)] = (short)f2bf(dvt1[r]);
    dkp[crow(r, hi)] = (short)f2bf(dkt0[r]);
    dkp[32 + crow(r, hi)] = (short)f2bf(dkt1[r]);
  }
}

// ============================================================================
// backward kernel B (q-parallel): dQ.  Lane owns one q row (the forward
// layout); dS is recomputed in the S^T layout and fed through the same
// permlane assembly as the forward's P.
//   dQ^T[d][q] += K^T[d][kv] @ dS^T[kv][q]
// ============================================================================
__global__ __launch_bounds__(256) void attn_bwd_dq_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dq, int64_t seq, float scale, int causal) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;
  const int64_t bh = blockIdx.y;
  const int64_t q0 = (int64_t)blockIdx.x * 128 + wave * 32;
  if (q0 >= seq) return;
  const short* qp = q + (bh * seq) * 64;
  const short* kp = k + (bh * seq) * 64;
  const short* vp = v + (bh * seq) * 64;
  const short* dop = dout + (bh * seq) * 64;

  const int64_t myq = q0 + lq;
  const int64_t qrow = myq < seq ? myq : seq - 1;
  const float mylse = lse[bh * seq + qrow];
  const float mydelta = delta[bh * seq + qrow];

  bf16x8 qfrag[4], dofrag[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    qfrag[c] = *reinterpret_cast<const bf16x8*>(
        qp + qrow * 64 + hi * 8 + 16 * c);
    dofrag[c] = *reinterpret_cast<const bf16x8*>(
        dop + qrow * 64 + hi * 8 + 16 * c);
  }

  f32x16 dqt0 = {}, dqt1 = {};   // dQ^T acc: rows d 0..31/32..63, col q

  const int64_t kv_end = causal ? (q0 + 32 < seq ? q0 + 32 : seq) : seq;
  for (int64_t kv0 = 0; kv0 < kv_end; kv0 += 32) {
    const int64_t kvrow = kv0 + lq < seq ? kv0 + lq : seq - 1;
    f32x16 st = {};
    f32x16 dpt = {};
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      // S^T = K @ Q^T (forward layout): A = K[m=kv][k=d], B = Q^T
      bf16x8 kf = *reinterpret_cast<const bf16x8*>(
          kp + kvrow * 64 + hi * 8 + 16 * c);
      bf16x8 vf = *reinterpret_cast<const bf16x8*>(
          vp + kvrow * 64 + hi * 8 + 16 * c);
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qfrag[c], st, 0, 0,
                                                   0);
      // dP^T = V @ dO^T: rows kv, col q
      dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dofrag[c], dpt, 0,
                                                    0, 0);
    }
    // st[r] = S[myq][kv0+crow(r,hi)]; dpt[r] = dP same layout
    float ds[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int64_t kvg = kv0 + crow(r, hi);
      bool masked = myq >= seq || kvg >= seq || (causal && kvg > myq);
      float pv = masked ? 0.f : __expf(st[r] * scale - mylse);
      ds[r] = masked ? 0.f : pv * (dpt[r] - mydelta) * scale;
    }
    bf16x8 db0 = assemble_pfrag(&ds[0]);   // dS^T[kv][q], kv chunk 0..15
    bf16x8 db1 = assemble_pfrag(&ds[8]);   // kv chunk 16..31
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      bf16x8 kt0, kt1;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int64_t kvg = kv0 + kc * 16 + hi * 8 + j;
        if (kvg >= seq) kvg = seq - 1;    // dS there is 0
        kt0[j] = kp[kvg * 64 + lq];
        kt1[j] = kp[kvg * 64 + 32 + lq];
      }
      bf16x8 db = kc == 0 ? db0 : db1;
      dqt0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kt0, db, dqt0, 0, 0,
                                                     0);
      dqt1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kt1, db, dqt1, 0, 0,
                                                     0);
    }
  }

  if (myq >= seq) return;
  short* dqp = dq + (bh * seq + myq) * 64;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    dqp[crow(r, hi)] = (short)f2bf(dqt0[r]);
    dqp[32 + crow(r, hi)] = (short)f2bf(dqt1[r]);
  }
}

__global__ void f32_to_bf16_4d_kernel(short* __restrict__ dst,
                                      const float* __restrict__ src,
                                      int64_t n) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dst[i] = (short)f2bf(src[i]);
}

}  // namespace

extern "C" {

void epl_attn_fwd(const void* q, const void* k, const void* v, void* out,
                  float* lse, int64_t batch_heads, int64_t seq, float scale,
                  bool causal, hipStream_t stream) {
  dim3 grid((unsigned)((seq + 127) / 128), (unsigned)batch_heads);
  hipLaunchKernelGGL(attn_fwd_kernel, grid, dim3(256), 0, stream,
                     reinterpret_cast<const short*>(q),
                     reinterpret_cast<const short*>(k),
                     reinterpret_cast<const short*>(v),
                     reinterpret_cast<short*>(out), lse, seq, scale,
                     causal ? 1 : 0);
}

void epl_attn_bwd(const void* q, const void* k, const void* v,
                  const void* out, const void* dout, const float* lse,
                  float* delta_ws, void* dq, void* dk, void* dv,
                  int64_t batch_heads, int64_t seq, float scale,
                  bool causal, hipStream_t stream) {
  const int64_t rows = batch_heads * seq;
  {
    const int64_t blocks = (rows + 3) / 4;
    hipLaunchKernelGGL(attn_bwd_prep_kernel,
                       dim3((unsigned)(blocks < 4096 ? blocks : 4096)),
                       dim3(256), 0, stream,
                       reinterpret_cast<const short*>(dout),
                       reinterpret_cast<const short*>(out), delta_ws, rows);
  }
  dim3 grid((unsigned)((seq + 127) / 128), (unsigned)batch_heads);
  hipLaunchKernelGGL(attn_bwd_dkdv_kernel, grid, dim3(256), 0, stream,
                     reinterpret_cast<const short*>(q),
                     reinterpret_cast<const short*>(k),
                     reinterpret_cast<const short*>(v),
                     reinterpret_cast<const short*>(dout), lse, delta_ws,
                     reinterpret_cast<short*>(dk),
                     reinterpret_cast<short*>(dv), seq, scale,
                     causal ? 1 : 0);
  hipLaunchKernelGGL(attn_bwd_dq_kernel, grid, dim3(256), 0, stream,
                     reinterpret_cast<const short*>(q),
                     reinterpret_cast<const short*>(k),
                     reinterpret_cast<const short*>(v),
                     reinterpret_cast<const short*>(dout), lse, delta_ws,
                     reinterpret_cast<short*>(dq), seq, scale,
                     causal ? 1 : 0);
}

}  // extern "C"
