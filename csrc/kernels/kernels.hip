// Hand-written CDNA4 (gfx950) kernels for the MI355X-native parallel library.
//
// These replace the hot-path ops the reference leaves to TF (the reference's
// native layer, /root/reference/csrc/, contains no device kernels at all —
// only NCCL glue).  Everything here is memory-bound streaming work, designed
// per the CDNA4 rules: 256-thread blocks (4 waves of 64), bf16 loaded as
// packed ushort vectors (8-16 B/lane), fp32 accumulation, grid-stride loops
// capped near 2048 blocks so the 256-CU chip is filled without launch spam,
// LDS used for per-block column-reduction partials (dgamma/dbeta/dbias) so
// global atomics fire once per block, not once per row.
//
// Optimizer kernels operate on FLAT ARENAS: the Python engine concatenates
// every parameter of a taskgraph into one contiguous buffer (master fp32 /
// model bf16 / grad), so one AdamW step is ONE kernel over one flat stream —
// no multi-tensor metadata tables, no per-parameter launches.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>
#include <cmath>

#define DEV __device__ __forceinline__

namespace {

constexpr int kBlock = 256;
constexpr int kMaxGrid = 4096;  // >> 256 CUs; grid-stride covers the rest

DEV float bf2f(unsigned short u) {
  unsigned int x = (unsigned int)u << 16;
  return __uint_as_float(x);
}

DEV unsigned short f2bf(float f) {
  // round-to-nearest-even bf16 conversion
  unsigned int x = __float_as_uint(f);
  unsigned int lsb = (x >> 16) & 1u;
  x += 0x7fffu + lsb;
  return (unsigned short)(x >> 16);
}

struct ushort8 {
  unsigned short v[8];
};
struct float8 {
  float v[8];
};

DEV float8 load_bf16x8(const unsigned short* p) {
  const ushort8 raw = *reinterpret_cast<const ushort8*>(p);
  float8 f;
#pragma unroll
  for (int i = 0; i < 8; ++i) f.v[i] = bf2f(raw.v[i]);
  return f;
}

DEV void store_bf16x8(unsigned short* p, const float8& f) {
  ushort8 raw;
#pragma unroll
  for (int i = 0; i < 8; ++i) raw.v[i] = f2bf(f.v[i]);
  *reinterpret_cast<ushort8*>(p) = raw;
}

DEV float8 load_f32x8(const float* p) {
  float8 f;
  const float4 a = *reinterpret_cast<const float4*>(p);
  const float4 b = *reinterpret_cast<const float4*>(p + 4);
  f.v[0] = a.x; f.v[1] = a.y; f.v[2] = a.z; f.v[3] = a.w;
  f.v[4] = b.x; f.v[5] = b.y; f.v[6] = b.z; f.v[7] = b.w;
  return f;
}

DEV void store_f32x8(float* p, const float8& f) {
  *reinterpret_cast<float4*>(p) = make_float4(f.v[0], f.v[1], f.v[2], f.v[3]);
  *reinterpret_cast<float4*>(p + 4) =
      make_float4(f.v[4], f.v[5], f.v[6], f.v[7]);
}

inline int grid_for(int64_t work_items) {
  int64_t blocks = (work_items + kBlock - 1) / kBlock;
  return (int)(blocks < kMaxGrid ? (blocks > 0 ? blocks : 1) : kMaxGrid);
}

// ---- block-wide reduction (256 threads = 4 waves) ---------------------------
DEV float block_reduce_sum(float val, float* lds /* >= 4 floats */) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    val += __shfl_down(val, off, 64);
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int nwaves = blockDim.x >> 6;
  if (lane == 0) lds[wave] = val;
  __syncthreads();
  if (wave == 0) {
    val = lane < nwaves ? lds[lane] : 0.f;
#pragma unroll
    for (int off = 2; off > 0; off >>= 1)
      val += __shfl_down(val, off, 64);
    if (lane == 0) lds[0] = val;
  }
  __syncthreads();
  return lds[0];
}

DEV float block_reduce_max(float val, float* lds) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    val = fmaxf(val, __shfl_down(val, off, 64));
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int nwaves = blockDim.x >> 6;
  if (lane == 0) lds[wave] = val;
  __syncthreads();
  if (wave == 0) {
    val = lane < nwaves ? lds[lane] : -INFINITY;
#pragma unroll
    for (int off = 2; off > 0; off >>= 1)
      val = fmaxf(val, __shfl_down(val, off, 64));
    if (lane == 0) lds[0] = val;
  }
  __syncthreads();
  return lds[0];
}

// ============================================================================
// Fused AdamW over a flat arena.
//   master: fp32 "true" parameters (flat).
//   param : optional bf16 working copy written back after the update.
//   grad  : bf16 or fp32 gradients; multiplied by inv_scale (AMP unscale
//           fused in).  Decoupled weight decay (AdamW).
// ============================================================================
template <bool GRAD_BF16, bool HAS_BF16_PARAM>
__global__ void fused_adamw_kernel(
    float* __restrict__ master, unsigned short* __restrict__ param,
    const void* __restrict__ grad_raw, float* __restrict__ m,
    float* __restrict__ v, int64_t n, float lr, float beta1, float beta2,
    float eps, float weight_decay, float inv_bias1, float inv_bias2,
    float inv_scale) {
  const int64_t nvec = n / 8;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t base = i * 8;
    float8 g;
    if (GRAD_BF16)
      g = load_bf16x8(
          reinterpret_cast<const unsigned short*>(grad_raw) + base);
    else
      g = load_f32x8(reinterpret_cast<const float*>(grad_raw) + base);
    float8 pm = load_f32x8(master + base);
    float8 mm = load_f32x8(m + base);
    float8 vv = load_f32x8(v + base);
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float gk = g.v[k] * inv_scale;
      float mk = beta1 * mm.v[k] + (1.f - beta1) * gk;
      float vk = beta2 * vv.v[k] + (1.f - beta2) * gk * gk;
      float mhat = mk * inv_bias1;
      float vhat = vk * inv_bias2;
      float pk = pm.v[k];
      pk -= lr * (mhat / (sqrtf(vhat) + eps) + weight_decay * pk);
      mm.v[k] = mk;
      vv.v[k] = vk;
      pm.v[k] = pk;
    }
    store_f32x8(m + base, mm);
    store_f32x8(v + base, vv);
    store_f32x8(master + base, pm);
    if (HAS_BF16_PARAM) store_bf16x8(param + base, pm);
  }
  // scalar tail
  const int64_t tail = nvec * 8;
  for (int64_t i = tail + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float gk = (GRAD_BF16
                    ? bf2f(reinterpret_cast<const unsigned short*>(grad_raw)[i])
                    : reinterpret_cast<const float*>(grad_raw)[i]) *
               inv_scale;
    float mk = beta1 * m[i] + (1.f - beta1) * gk;
    float vk = beta2 * v[i] + (1.f - beta2) * gk * gk;
    float pk = master[i];
    pk -= lr * ((mk * inv_bias1) / (sqrtf(vk * inv_bias2) + eps) +
                weight_decay * pk);
    m[i] = mk;
    v[i] = vk;
    master[i] = pk;
    if (HAS_BF16_PARAM) param[i] = f2bf(pk);
  }
}

// ============================================================================
// Fused LAMB over a flat arena, phase kernels.
//   Phase 1: update m, v; write the raw Adam update u into a scratch buffer;
//            accumulate per-CHUNK squared norms of (param, update) so the
//            Python side can compute per-parameter trust ratios (chunks are
//            parameter boundaries in the flat arena).
//   Phase 2: p -= lr * ratio[chunk] * (u + wd*p) applied per chunk.
// Chunk ids come from a per-8-element chunk-index map built once on the host.
// ============================================================================
__global__ void lamb_phase1_kernel(
    const float* __restrict__ master, const void* __restrict__ grad_raw,
    bool grad_bf16, float* __restrict__ m, float* __restrict__ v,
    float* __restrict__ update, const int* __restrict__ chunk_of,
    float* __restrict__ wnorm_sq, float* __restrict__ unorm_sq, int64_t n,
    float beta1, float beta2, float eps, float weight_decay, float inv_bias1,
    float inv_bias2, float inv_scale) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float gk = (grad_bf16
                    ? bf2f(reinterpret_cast<const unsigned short*>(grad_raw)[i])
                    : reinterpret_cast<const float*>(grad_raw)[i]) *
               inv_scale;
    float mk = beta1 * m[i] + (1.f - beta1) * gk;
    float vk = beta2 * v[i] + (1.f - beta2) * gk * gk;
    m[i] = mk;
    v[i] = vk;
    float pk = master[i];
    float u = (mk * inv_bias1) / (sqrtf(vk * inv_bias2) + eps) +
              weight_decay * pk;
    update[i] = u;
    const int c = chunk_of[i >> 3];
    atomicAdd(&wnorm_sq[c], pk * pk);
    atomicAdd(&unorm_sq[c], u * u);
  }
}

__global__ void lamb_phase2_kernel(float* __restrict__ master,
                                   unsigned short* __restrict__ param,
                                   const float* __restrict__ update,
                                   const int* __restrict__ chunk_of,
                                   const float* __restrict__ ratio, int64_t n,
                                   float lr) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int c = chunk_of[i >> 3];
    float pk = master[i] - lr * ratio[c] * update[i];
    master[i] = pk;
    if (param != nullptr) param[i] = f2bf(pk);
  }
}

// ============================================================================
// LayerNorm forward with optional fused residual add: s = x (+ res);
// y = (s - mean(s)) * rstd(s) * gamma + beta.  One block per row.  The
// bf16 fast path caches each thread's own row values in REGISTERS between
// the statistics pass and the normalize pass (threads re-touch the same
// columns), so HBM is read once and LDS carries only the 8-float block
// reduction.  CHUNKS = row float8-chunks per thread; the launcher picks
// the block size so CHUNKS <= 2 covers cols <= 4096 at 256 threads.
// With HAS_RES the summed input s is also written out (the tensor the
// backward normalizes against and, for pre-LN blocks, the residual
// stream).  Saves mean and rstd.
// ============================================================================
template <bool HAS_RES, int CHUNKS>
__global__ void layer_norm_fwd_bf16_kernel(
    void* __restrict__ out, const void* __restrict__ x,
    const void* __restrict__ res, void* __restrict__ sum_out,
    const void* __restrict__ gamma, const void* __restrict__ beta,
    float* __restrict__ mean_out, float* __restrict__ rstd_out, int64_t rows,
    int64_t cols, float eps) {
  __shared__ float lds[8];
  using T = unsigned short;
  float8 cache[CHUNKS];
  float8 gw[CHUNKS], bw[CHUNKS];
#pragma unroll
  for (int k = 0; k < CHUNKS; ++k) {
    const int64_t c = ((int64_t)threadIdx.x + k * blockDim.x) * 8;
    if (c < cols) {
      gw[k] = load_bf16x8(reinterpret_cast<const T*>(gamma) + c);
      bw[k] = load_bf16x8(reinterpret_cast<const T*>(beta) + c);
    }
  }
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const int64_t base = row * cols;
    float sum = 0.f, sumsq = 0.f;
#pragma unroll
    for (int k = 0; k < CHUNKS; ++k) {
      const int64_t c = ((int64_t)threadIdx.x + k * blockDim.x) * 8;
      if (c >= cols) continue;
      float8 f = load_bf16x8(reinterpret_cast<const T*>(x) + base + c);
      if (HAS_RES) {
        float8 r = load_bf16x8(reinterpret_cast<const T*>(res) + base + c);
#pragma unroll
        for (int j = 0; j < 8; ++j) f.v[j] += r.v[j];
        store_bf16x8(reinterpret_cast<T*>(sum_out) + base + c, f);
      }
      cache[k] = f;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        sum += f.v[j];
        sumsq += f.v[j] * f.v[j];
      }
    }
    sum = block_reduce_sum(sum, lds);
    __syncthreads();
    sumsq = block_reduce_sum(sumsq, lds);
    const float mean = sum / cols;
    const float var = sumsq / cols - mean * mean;
    const float rstd = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
#pragma unroll
    for (int k = 0; k < CHUNKS; ++k) {
      const int64_t c = ((int64_t)threadIdx.x + k * blockDim.x) * 8;
      if (c >= cols) continue;
      float8 y;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        y.v[j] = (cache[k].v[j] - mean) * rstd * gw[k].v[j] + bw[k].v[j];
      store_bf16x8(reinterpret_cast<T*>(out) + base + c, y);
    }
    __syncthreads();
  }
}

// generic (fp32 / large-cols) forward: two-pass re-read
template <bool BF16, bool HAS_RES>
__global__ void layer_norm_fwd_kernel(
    void* __restrict__ out, const void* __restrict__ x,
    const void* __restrict__ res, void* __restrict__ sum_out,
    const void* __restrict__ gamma, const void* __restrict__ beta,
    float* __restrict__ mean_out, float* __restrict__ rstd_out, int64_t rows,
    int64_t cols, float eps) {
  __shared__ float lds[8];
  using T = unsigned short;
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const int64_t base = row * cols;
    float sum = 0.f, sumsq = 0.f;
    for (int64_t c = threadIdx.x; c < cols; c += blockDim.x) {
      float fv = BF16 ? bf2f(reinterpret_cast<const T*>(x)[base + c])
                      : reinterpret_cast<const float*>(x)[base + c];
      if (HAS_RES) {
        fv += BF16 ? bf2f(reinterpret_cast<const T*>(res)[base + c])
                   : reinterpret_cast<const float*>(res)[base + c];
        if (BF16)
          reinterpret_cast<T*>(sum_out)[base + c] = f2bf(fv);
        else
          reinterpret_cast<float*>(sum_out)[base + c] = fv;
      }
      sum += fv;
      sumsq += fv * fv;
    }
    sum = block_reduce_sum(sum, lds);
    __syncthreads();
    sumsq = block_reduce_sum(sumsq, lds);
    const float mean = sum / cols;
    const float var = sumsq / cols - mean * mean;
    const float rstd = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    const void* nin = HAS_RES ? sum_out : x;
    for (int64_t c = threadIdx.x; c < cols; c += blockDim.x) {
      float fv = BF16 ? bf2f(reinterpret_cast<const T*>(nin)[base + c])
                      : reinterpret_cast<const float*>(nin)[base + c];
      float gv = BF16 ? bf2f(reinterpret_cast<const T*>(gamma)[c])
                      : reinterpret_cast<const float*>(gamma)[c];
      float bv = BF16 ? bf2f(reinterpret_cast<const T*>(beta)[c])
                      : reinterpret_cast<const float*>(beta)[c];
      float y = (fv - mean) * rstd * gv + bv;
      if (BF16)
        reinterpret_cast<T*>(out)[base + c] = f2bf(y);
      else
        reinterpret_cast<float*>(out)[base + c] = y;
    }
    __syncthreads();
  }
}

// bf16 fast backward: ONE ROW PER WAVE — no barriers at all.  Each wave
// reduces its row with shuffles, keeps the row image in registers between
// the reduction and the dx pass (HBM read-once/write-once), accumulates
// dgamma/dbeta in registers across its rows and flushes once with
// atomics.  CHUNKS = ceil(cols/8/64); fast path cols <= 1024 (CHUNKS<=2).
template <int CHUNKS, bool CACHE>
__global__ __launch_bounds__(256, 2) void layer_norm_bwd_bf16_kernel(
    void* __restrict__ dx, float* __restrict__ dg_part,
    float* __restrict__ db_part, const void* __restrict__ dy,
    const void* __restrict__ x, const void* __restrict__ gamma,
    const float* __restrict__ mean_in, const float* __restrict__ rstd_in,
    int64_t rows, int64_t cols) {
  using T = unsigned short;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;               // wave in block (0..3)
  const int wpb = blockDim.x >> 6;                // waves per block
  float8 gw[CHUNKS], acc_dg[CHUNKS], acc_db[CHUNKS];
#pragma unroll
  for (int k = 0; k < CHUNKS; ++k) {
    const int64_t c = ((int64_t)lane + k * 64) * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      acc_dg[k].v[j] = 0.f;
      acc_db[k].v[j] = 0.f;
    }
    if (c < cols)
      gw[k] = load_bf16x8(reinterpret_cast<const T*>(gamma) + c);
  }
  for (int64_t row = (int64_t)blockIdx.x * wpb + wid; row < rows;
       row += (int64_t)gridDim.x * wpb) {
    const int64_t base = row * cols;
    const float mean = mean_in[row];
    const float rstd = rstd_in[row];
    float sum_dyg = 0.f, sum_dygx = 0.f;
    float8 c_xhat[CACHE ? CHUNKS : 1], c_dy[CACHE ? CHUNKS : 1];
#pragma unroll
    for (int k = 0; k < CHUNKS; ++k) {
      const int64_t c = ((int64_t)lane + k * 64) * 8;
      if (c >= cols) continue;
      float8 d = load_bf16x8(reinterpret_cast<const T*>(dy) + base + c);
      float8 f = load_bf16x8(reinterpret_cast<const T*>(x) + base + c);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float xhat = (f.v[j] - mean) * rstd;
        const float dyg = d.v[j] * gw[k].v[j];
        if (CACHE) {
          c_xhat[k].v[j] = xhat;
          c_dy[k].v[j] = d.v[j];
        }
        sum_dyg += dyg;
        sum_dygx += dyg * xhat;
      }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      sum_dyg += __shfl_down(sum_dyg, off, 64);
      sum_dygx += __shfl_down(sum_dygx, off, 64);
    }
    sum_dyg = __shfl(sum_dyg, 0, 64);
    sum_dygx = __shfl(sum_dygx, 0, 64);
    const float inv_cols = 1.f / cols;
#pragma unroll
    for (int k = 0; k < CHUNKS; ++k) {
      const int64_t c = ((int64_t)lane + k * 64) * 8;
      if (c >= cols) continue;
      float8 dv, xh;
      if (CACHE) {
        dv = c_dy[k];
        xh = c_xhat[k];
      } else {
        // rows in flight fit L2: the re-reads hit cache
        float8 d = load_bf16x8(reinterpret_cast<const T*>(dy) + base + c);
        float8 f = load_bf16x8(reinterpret_cast<const T*>(x) + base + c);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          dv.v[j] = d.v[j];
          xh.v[j] = (f.v[j] - mean) * rstd;
        }
      }
      float8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float dyg = dv.v[j] * gw[k].v[j];
        o.v[j] = (dyg - (sum_dyg + xh.v[j] * sum_dygx) * inv_cols) * rstd;
        acc_dg[k].v[j] += dv.v[j] * xh.v[j];
        acc_db[k].v[j] += dv.v[j];
      }
      store_bf16x8(reinterpret_cast<T*>(dx) + base + c, o);
    }
  }
  // one partial row per wave — no atomics, a tiny second kernel reduces
  const int64_t prow = ((int64_t)blockIdx.x * wpb + wid) * cols;
#pragma unroll
  for (int k = 0; k < CHUNKS; ++k) {
    const int64_t c = ((int64_t)lane + k * 64) * 8;
    if (c >= cols) continue;
    store_f32x8(dg_part + prow + c, acc_dg[k]);
    store_f32x8(db_part + prow + c, acc_db[k]);
  }
}

// reduce the [nparts, cols] partial buffers into dgamma/dbeta.
// grid = (col_chunks, PSPLIT): each block sums a strided slice of the
// partial rows for its 256-column range (coalesced row-major reads) and
// contributes one atomicAdd per column (PSPLIT per column total).
__global__ void ln_bwd_partial_reduce_kernel(
    float* __restrict__ dgamma, float* __restrict__ dbeta,
    const float* __restrict__ dg_part, const float* __restrict__ db_part,
    int64_t nparts, int64_t cols) {
  const int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= cols) return;
  float sg = 0.f, sb = 0.f;
  for (int64_t p = blockIdx.y; p < nparts; p += gridDim.y) {
    sg += dg_part[p * cols + c];
    sb += db_part[p * cols + c];
  }
  atomicAdd(&dgamma[c], sg);
  atomicAdd(&dbeta[c], sb);
}

// generic (fp32 / large-cols) backward: LDS partials + global re-read
template <bool BF16>
__global__ void layer_norm_bwd_kernel(
    void* __restrict__ dx, float* __restrict__ dgamma,
    float* __restrict__ dbeta, const void* __restrict__ dy,
    const void* __restrict__ x, const void* __restrict__ gamma,
    const float* __restrict__ mean_in, const float* __restrict__ rstd_in,
    int64_t rows, int64_t cols) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* part_dg = reinterpret_cast<float*>(smem);            // [cols]
  float* part_db = part_dg + cols;                            // [cols]
  __shared__ float lds[8];
  using T = unsigned short;
  for (int64_t c = threadIdx.x; c < cols; c += blockDim.x) {
    part_dg[c] = 0.f;
    part_db[c] = 0.f;
  }
  __syncthreads();
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const int64_t base = row * cols;
    const float mean = mean_in[row];
    const float rstd = rstd_in[row];
    float sum_dyg = 0.f, sum_dygx = 0.f;
    for (int64_t c = threadIdx.x; c < cols; c += blockDim.x) {
      const float dv = BF16 ? bf2f(reinterpret_cast<const T*>(dy)[base + c])
                            : reinterpret_cast<const float*>(dy)[base + c];
      const float fv = BF16 ? bf2f(reinterpret_cast<const T*>(x)[base + c])
                            : reinterpret_cast<const float*>(x)[base + c];
      const float gv = BF16 ? bf2f(reinterpret_cast<const T*>(gamma)[c])
                            : reinterpret_cast<const float*>(gamma)[c];
      const float xhat = (fv - mean) * rstd;
      const float dyg = dv * gv;
      sum_dyg += dyg;
      sum_dygx += dyg * xhat;
      part_dg[c] += dv * xhat;
      part_db[c] += dv;
    }
    __syncthreads();
    sum_dyg = block_reduce_sum(sum_dyg, lds);
    __syncthreads();
    sum_dygx = block_reduce_sum(sum_dygx, lds);
    const float inv_cols = 1.f / cols;
    for (int64_t c = threadIdx.x; c < cols; c += blockDim.x) {
      const float dv = BF16 ? bf2f(reinterpret_cast<const T*>(dy)[base + c])
                            : reinterpret_cast<const float*>(dy)[base + c];
      const float fv = BF16 ? bf2f(reinterpret_cast<const T*>(x)[base + c])
                            : reinterpret_cast<const float*>(x)[base + c];
      const float gv = BF16 ? bf2f(reinterpret_cast<const T*>(gamma)[c])
                            : reinterpret_cast<const float*>(gamma)[c];
      const float xhat = (fv - mean) * rstd;
      const float o = (dv * gv - (sum_dyg + xhat * sum_dygx) * inv_cols) *
                      rstd;
      if (BF16)
        reinterpret_cast<T*>(dx)[base + c] = f2bf(o);
      else
        reinterpret_cast<float*>(dx)[base + c] = o;
    }
    __syncthreads();
  }
  for (int64_t c = threadIdx.x; c < cols; c += blockDim.x) {
    atomicAdd(&dgamma[c], part_dg[c]);
    atomicAdd(&dbeta[c], part_db[c]);
  }
}

// ============================================================================
// Fused bias + GeLU (tanh approximation), forward and backward.
// ============================================================================
DEV float fast_tanh(float u) {
  // tanh(u) = 1 - 2/(e^2u + 1): one fast exp instead of tanhf's polynomial
  // + range-reduction chain (bf16-accurate; numerics-tested vs torch)
  return 1.f - 2.f / (__expf(2.f * u) + 1.f);
}

DEV float gelu_f(float x) {
  const float k0 = 0.7978845608028654f;   // sqrt(2/pi)
  const float k1 = 0.044715f;
  return 0.5f * x * (1.f + fast_tanh(k0 * (x + k1 * x * x * x)));
}

DEV float gelu_grad_f(float x) {
  const float k0 = 0.7978845608028654f;
  const float k1 = 0.044715f;
  const float t = fast_tanh(k0 * (x + k1 * x * x * x));
  const float dt = (1.f - t * t) * k0 * (1.f + 3.f * k1 * x * x);
  return 0.5f * (1.f + t) + 0.5f * x * dt;
}

template <bool BF16>
__global__ void bias_gelu_fwd_kernel(void* __restrict__ out,
                                     const void* __restrict__ x,
                                     const void* __restrict__ bias,
                                     int64_t rows, int64_t cols) {
  using T = unsigned short;
  const int64_t n = rows * cols;
  if (BF16) {
    const int64_t nvec = n / 8;
    for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
         i += (int64_t)gridDim.x * blockDim.x) {
      const int64_t base = i * 8;
      float8 f = load_bf16x8(reinterpret_cast<const T*>(x) + base);
      float8 b = load_bf16x8(reinterpret_cast<const T*>(bias) +
                             (base % cols));
      float8 o;
#pragma unroll
      for (int k = 0; k < 8; ++k) o.v[k] = gelu_f(f.v[k] + b.v[k]);
      store_bf16x8(reinterpret_cast<T*>(out) + base, o);
    }
  } else {
    for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (int64_t)gridDim.x * blockDim.x) {
      const float xv = reinterpret_cast<const float*>(x)[i] +
                       reinterpret_cast<const float*>(bias)[i % cols];
      reinterpret_cast<float*>(out)[i] = gelu_f(xv);
    }
  }
}

// bf16 fast path: one row-segment per wave, dbias accumulated in
// registers across rows and flushed to per-wave partial rows (the LN
// partial-reduce kernel pattern).  SEGS = row segments of 512 columns;
// each wave owns one 512-col segment across ALL rows so its dbias
// registers cover exactly its columns.
__global__ void bias_gelu_bwd_bf16_kernel(
    void* __restrict__ dx, float* __restrict__ db_part,
    const void* __restrict__ dy, const void* __restrict__ x,
    const void* __restrict__ bias, int64_t rows, int64_t cols) {
  using T = unsigned short;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = blockDim.x >> 6;
  const int64_t segs = cols / 512;              // cols % 512 == 0
  const int64_t total = rows * segs;
  float8 acc;
#pragma unroll
  for (int j = 0; j < 8; ++j) acc.v[j] = 0.f;
  const int64_t wave_id = (int64_t)blockIdx.x * wpb + wid;
  const int64_t nwaves = (int64_t)gridDim.x * wpb;
  // walk (row, my fixed segment) pairs: wave w owns segment w % segs
  const int64_t myseg = wave_id % segs;
  const int64_t c0 = myseg * 512 + (int64_t)lane * 8;
  float8 bw = load_bf16x8(reinterpret_cast<const T*>(bias) + c0);
  for (int64_t row = wave_id / segs; row < rows; row += nwaves / segs) {
    const int64_t base = row * cols + c0;
    float8 d = load_bf16x8(reinterpret_cast<const T*>(dy) + base);
    float8 f = load_bf16x8(reinterpret_cast<const T*>(x) + base);
    float8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float g = d.v[j] * gelu_grad_f(f.v[j] + bw.v[j]);
      o.v[j] = g;
      acc.v[j] += g;
    }
    store_bf16x8(reinterpret_cast<T*>(dx) + base, o);
  }
  store_f32x8(db_part + wave_id * 512 + (int64_t)lane * 8, acc);
}

// reduce [nwaves, 512-per-seg...] — reuse ln partial reduce over a
// [nparts, cols] view is not directly applicable; dbias partials are
// [nwaves][512] keyed by segment: reduce per (segment, col) over the
// waves owning that segment.
__global__ void bias_gelu_db_reduce_kernel(
    float* __restrict__ dbias, const float* __restrict__ db_part,
    int64_t nwaves, int64_t segs) {
  // column c (global) = seg * 512 + off; partial p covers segment
  // p % segs at its [512] row.
  const int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= segs * 512) return;
  const int64_t seg = c / 512;
  const int64_t off = c % 512;
  float sum = 0.f;
  for (int64_t p = seg + blockIdx.y * segs; p < nwaves;
       p += (int64_t)gridDim.y * segs) {
    sum += db_part[p * 512 + off];
  }
  atomicAdd(&dbias[c], sum);
}

template <bool BF16>
__global__ void bias_gelu_bwd_kernel(void* __restrict__ dx,
                                     float* __restrict__ dbias,
                                     const void* __restrict__ dy,
                                     const void* __restrict__ x,
                                     const void* __restrict__ bias,
                                     int64_t rows, int64_t cols) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* part_db = reinterpret_cast<float*>(smem);  // [cols]
  using T = unsigned short;
  for (int64_t c = threadIdx.x; c < cols; c += blockDim.x) part_db[c] = 0.f;
  __syncthreads();
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const int64_t base = row * cols;
    if (BF16) {
      for (int64_t c = threadIdx.x * 8; c < cols; c += (int64_t)blockDim.x * 8) {
        float8 d = load_bf16x8(reinterpret_cast<const T*>(dy) + base + c);
        float8 f = load_bf16x8(reinterpret_cast<const T*>(x) + base + c);
        float8 b = load_bf16x8(reinterpret_cast<const T*>(bias) + c);
        float8 o;
#pragma unroll
        for (int k = 0; k < 8; ++k) {
          const float g = d.v[k] * gelu_grad_f(f.v[k] + b.v[k]);
          o.v[k] = g;
          part_db[c + k] += g;
        }
        store_bf16x8(reinterpret_cast<T*>(dx) + base + c, o);
      }
    } else {
      for (int64_t c = threadIdx.x; c < cols; c += blockDim.x) {
        const float g = reinterpret_cast<const float*>(dy)[base + c] *
                        gelu_grad_f(reinterpret_cast<const float*>(x)[base + c] +
                                    reinterpret_cast<const float*>(bias)[c]);
        reinterpret_cast<float*>(dx)[base + c] = g;
        part_db[c] += g;
      }
    }
  }
  __syncthreads();
  for (int64_t c = threadIdx.x; c < cols; c += blockDim.x)
    atomicAdd(&dbias[c], part_db[c]);
}

// ============================================================================
// Fused softmax cross-entropy over (possibly vocab-sharded) logits.
// Forward: per row, max + sum(exp) in one pass each; writes per-row
// (max, sumexp, loss-contribution).  For the TP-sharded case the Python side
// allreduces max/sumexp between the two phases (reference semantics:
// /root/reference/epl/ops/distributed_losses.py:58-109).
//   Phase A (local): row max and sum exp(x - rowmax), plus the local logit at
//   the target (if the target falls in this shard's [vocab_begin, end)).
//   Phase B (local, after reduction): dlogits = softmax - onehot.
// ============================================================================
template <bool BF16>
__global__ void ce_rowstats_kernel(const void* __restrict__ logits,
                                   const int64_t* __restrict__ targets,
                                   float* __restrict__ row_max,
                                   float* __restrict__ row_sumexp,
                                   float* __restrict__ target_logit,
                                   int64_t rows, int64_t cols,
                                   int64_t vocab_begin, int64_t ignore_index) {
  __shared__ float lds[8];
  using T = unsigned short;
  const bool vec = BF16 && (cols % 8 == 0);
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const int64_t base = row * cols;
    float vmax = -INFINITY;
    if (vec) {
      const T* lr = reinterpret_cast<const T*>(logits) + base;
      for (int64_t c = threadIdx.x * 8; c < cols;
           c += (int64_t)blockDim.x * 8) {
        float8 f = load_bf16x8(lr + c);
#pragma unroll
        for (int k = 0; k < 8; ++k) vmax = fmaxf(vmax, f.v[k]);
      }
    } else {
      for (int64_t c = threadIdx.x; c < cols; c += blockDim.x) {
        const float v = BF16
            ? bf2f(reinterpret_cast<const T*>(logits)[base + c])
            : reinterpret_cast<const float*>(logits)[base + c];
        vmax = fmaxf(vmax, v);
      }
    }
    vmax = block_reduce_max(vmax, lds);
    __syncthreads();
    float sum = 0.f;
    if (vec) {
      const T* lr = reinterpret_cast<const T*>(logits) + base;
      for (int64_t c = threadIdx.x * 8; c < cols;
           c += (int64_t)blockDim.x * 8) {
        float8 f = load_bf16x8(lr + c);
#pragma unroll
        for (int k = 0; k < 8; ++k) sum += expf(f.v[k] - vmax);
      }
    } else {
      for (int64_t c = threadIdx.x; c < cols; c += blockDim.x) {
        const float v = BF16
            ? bf2f(reinterpret_cast<const T*>(logits)[base + c])
            : reinterpret_cast<const float*>(logits)[base + c];
        sum += expf(v - vmax);
      }
    }
    sum = block_reduce_sum(sum, lds);
    if (threadIdx.x == 0) {
      row_max[row] = vmax;
      // raw sum at this shard's max; Python rescales by
      // exp(local_max - global_max) before summing across shards.
      row_sumexp[row] = sum;
      const int64_t tgt = targets[row];
      float tl = 0.f;
      if (tgt != ignore_index && tgt >= vocab_begin &&
          tgt < vocab_begin + cols) {
        const int64_t c = tgt - vocab_begin;
        tl = BF16 ? bf2f(reinterpret_cast<const T*>(logits)[base + c])
                  : reinterpret_cast<const float*>(logits)[base + c];
      }
      target_logit[row] = tl;
    }
    __syncthreads();
  }
}

template <bool BF16>
__global__ void ce_backward_kernel(void* __restrict__ dlogits,
                                   const void* __restrict__ logits,
                                   const int64_t* __restrict__ targets,
                                   const float* __restrict__ gmax,
                                   const float* __restrict__ gsumexp,
                                   const float* __restrict__ dloss,
                                   int64_t rows, int64_t cols,
                                   int64_t vocab_begin, int64_t ignore_index,
                                   float scale) {
  using T = unsigned short;
  const bool vec = BF16 && (cols % 8 == 0);
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const int64_t base = row * cols;
    const float m = gmax[row];
    const float inv_sum = 1.f / gsumexp[row];
    const int64_t tgt = targets[row];
    const float dl = (tgt == ignore_index ? 0.f : dloss[row] * scale);
    const int64_t tc = tgt - vocab_begin;
    if (vec) {
      const T* lr = reinterpret_cast<const T*>(logits) + base;
      T* dr = reinterpret_cast<T*>(dlogits) + base;
      for (int64_t c = threadIdx.x * 8; c < cols;
           c += (int64_t)blockDim.x * 8) {
        float8 f = load_bf16x8(lr + c);
        float8 o;
#pragma unroll
        for (int k = 0; k < 8; ++k) {
          float g = expf(f.v[k] - m) * inv_sum;
          if (tc == c + k) g -= 1.f;
          o.v[k] = g * dl;
        }
        store_bf16x8(dr + c, o);
      }
    } else {
      for (int64_t c = threadIdx.x; c < cols; c += blockDim.x) {
        const float v = BF16
            ? bf2f(reinterpret_cast<const T*>(logits)[base + c])
            : reinterpret_cast<const float*>(logits)[base + c];
        float g = expf(v - m) * inv_sum;
        if (tc == c) g -= 1.f;
        g *= dl;
        if (BF16)
          reinterpret_cast<T*>(dlogits)[base + c] = f2bf(g);
        else
          reinterpret_cast<float*>(dlogits)[base + c] = g;
      }
    }
  }
}

// ============================================================================
// Flat-arena utilities: zero / scale / axpy / casts / sq-norm.
// ============================================================================
__global__ void scale_f32_kernel(float* __restrict__ p, int64_t n, float s) {
  const int64_t nvec = n / 4;
  float4* p4 = reinterpret_cast<float4*>(p);
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    float4 v = p4[i];
    v.x *= s; v.y *= s; v.z *= s; v.w *= s;
    p4[i] = v;
  }
  for (int64_t i = nvec * 4 + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    p[i] *= s;
}

__global__ void scale_bf16_kernel(unsigned short* __restrict__ p, int64_t n,
                                  float s) {
  const int64_t nvec = n / 8;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    float8 v = load_bf16x8(p + i * 8);
#pragma unroll
    for (int k = 0; k < 8; ++k) v.v[k] *= s;
    store_bf16x8(p + i * 8, v);
  }
  for (int64_t i = nvec * 8 + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    p[i] = f2bf(bf2f(p[i]) * s);
}

__global__ void f32_to_bf16_kernel(unsigned short* __restrict__ dst,
                                   const float* __restrict__ src, int64_t n) {
  const int64_t nvec = n / 8;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x)
    store_bf16x8(dst + i * 8, load_f32x8(src + i * 8));
  for (int64_t i = nvec * 8 + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dst[i] = f2bf(src[i]);
}

__global__ void bf16_to_f32_kernel(float* __restrict__ dst,
                                   const unsigned short* __restrict__ src,
                                   int64_t n) {
  const int64_t nvec = n / 8;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x)
    store_f32x8(dst + i * 8, load_bf16x8(src + i * 8));
  for (int64_t i = nvec * 8 + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dst[i] = bf2f(src[i]);
}

template <bool BF16>
__global__ void sqnorm_kernel(const void* __restrict__ p, int64_t n,
                              float* __restrict__ out) {
  __shared__ float lds[8];
  using T = unsigned short;
  float acc = 0.f;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const float v = BF16 ? bf2f(reinterpret_cast<const T*>(p)[i])
                         : reinterpret_cast<const float*>(p)[i];
    acc += v * v;
  }
  acc = block_reduce_sum(acc, lds);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

}  // namespace

// ============================================================================

// ============================================================================
// Column sum (Linear bias gradient): db[c] = sum_r dy[r, c].
// torch's generic reduce runs at ~2.2 TB/s on [65536, 1024] bf16; this
// pair streams bf16x8 rows into fp32 stripe partials (no atomics in the
// hot pass, same pattern as the LN backward) and a tiny reduce kernel
// folds the stripes.  ~73 of these per BERT-Large step (qkv/proj/fc2
// biases).
// ============================================================================
template <bool BF16>
__global__ void colsum_partial_kernel(const void* __restrict__ dy,
                                      float* __restrict__ partial,
                                      int64_t rows, int64_t cols,
                                      int64_t stripes) {
  // grid: (ceil(cols/(256*8)), stripes); each thread owns 8 columns
  const int64_t c = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (c >= cols) return;
  const int64_t per = (rows + stripes - 1) / stripes;
  const int64_t r0 = (int64_t)blockIdx.y * per;
  const int64_t r1 = r0 + per < rows ? r0 + per : rows;
  float8 acc = {};
  for (int64_t r = r0; r < r1; ++r) {
    if (BF16) {
      const float8 v =
          load_bf16x8(reinterpret_cast<const unsigned short*>(dy) +
                      r * cols + c);
#pragma unroll
      for (int k = 0; k < 8; ++k) acc.v[k] += v.v[k];
    } else {
      const float* p = reinterpret_cast<const float*>(dy) + r * cols + c;
#pragma unroll
      for (int k = 0; k < 8; ++k) acc.v[k] += p[k];
    }
  }
  store_f32x8(partial + (int64_t)blockIdx.y * cols + c, acc);
}

template <bool BF16>
__global__ void colsum_reduce_kernel(void* __restrict__ db,
                                     const float* __restrict__ partial,
                                     int64_t stripes, int64_t cols) {
  const int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= cols) return;
  float s = 0.f;
  for (int64_t p = 0; p < stripes; ++p) s += partial[p * cols + c];
  if (BF16)
    reinterpret_cast<unsigned short*>(db)[c] = f2bf(s);
  else
    reinterpret_cast<float*>(db)[c] = s;
}

// extern "C" launchers (raw pointers + stream; bound to torch in bindings.hip)
// ============================================================================
extern "C" {

void epl_fused_adamw(float* master, unsigned short* param_bf16,
                     const void* grad, bool grad_bf16, float* m, float* v,
                     int64_t n, float lr, float beta1, float beta2, float eps,
                     float weight_decay, int step, float inv_scale,
                     hipStream_t stream) {
  const float inv_bias1 = 1.f / (1.f - powf(beta1, (float)step));
  const float inv_bias2 = 1.f / (1.f - powf(beta2, (float)step));
  const int grid = grid_for((n + 7) / 8);
  if (grad_bf16) {
    if (param_bf16)
      hipLaunchKernelGGL((fused_adamw_kernel<true, true>), dim3(grid),
                         dim3(kBlock), 0, stream, master, param_bf16, grad, m,
                         v, n, lr, beta1, beta2, eps, weight_decay, inv_bias1,
                         inv_bias2, inv_scale);
    else
      hipLaunchKernelGGL((fused_adamw_kernel<true, false>), dim3(grid),
                         dim3(kBlock), 0, stream, master, param_bf16, grad, m,
                         v, n, lr, beta1, beta2, eps, weight_decay, inv_bias1,
                         inv_bias2, inv_scale);
  } else {
    if (param_bf16)
      hipLaunchKernelGGL((fused_adamw_kernel<false, true>), dim3(grid),
                         dim3(kBlock), 0, stream, master, param_bf16, grad, m,
                         v, n, lr, beta1, beta2, eps, weight_decay, inv_bias1,
                         inv_bias2, inv_scale);
    else
      hipLaunchKernelGGL((fused_adamw_kernel<false, false>), dim3(grid),
                         dim3(kBlock), 0, stream, master, param_bf16, grad, m,
                         v, n, lr, beta1, beta2, eps, weight_decay, inv_bias1,
                         inv_bias2, inv_scale);
  }
}

void epl_lamb_phase1(const float* master, const void* grad, bool grad_bf16,
                     float* m, float* v, float* update, const int* chunk_of,
                     float* wnorm_sq, float* unorm_sq, int64_t n, float beta1,
                     float beta2, float eps, float weight_decay, int step,
                     float inv_scale, hipStream_t stream) {
  const float inv_bias1 = 1.f / (1.f - powf(beta1, (float)step));
  const float inv_bias2 = 1.f / (1.f - powf(beta2, (float)step));
  const int grid = grid_for(n);
  hipLaunchKernelGGL(lamb_phase1_kernel, dim3(grid), dim3(kBlock), 0, stream,
                     master, grad, grad_bf16, m, v, update, chunk_of, wnorm_sq,
                     unorm_sq, n, beta1, beta2, eps, weight_decay, inv_bias1,
                     inv_bias2, inv_scale);
}

void epl_lamb_phase2(float* master, unsigned short* param_bf16,
                     const float* update, const int* chunk_of,
                     const float* ratio, int64_t n, float lr,
                     hipStream_t stream) {
  const int grid = grid_for(n);
  hipLaunchKernelGGL(lamb_phase2_kernel, dim3(grid), dim3(kBlock), 0, stream,
                     master, param_bf16, update, chunk_of, ratio, n, lr);
}

static int ln_block_for(int64_t cols) {
  // maximize threads per row: CHUNKS=1 whenever cols/8 fits in <=256
  // threads; CHUNKS=2 only for cols in (2048, 4096]
  int64_t need = (cols + 7) / 8;          // float8 chunks in a row
  if (need <= 64) return 64;
  if (need <= 128) return 128;
  if (need <= 192) return 192;
  return 256;
}

void epl_layer_norm_fwd(void* out, const void* x, const void* res,
                        void* sum_out, const void* gamma, const void* beta,
                        float* mean, float* rstd, int64_t rows, int64_t cols,
                        float eps, bool bf16, hipStream_t stream) {
  const int grid = (int)(rows < kMaxGrid ? rows : kMaxGrid);
  const bool has_res = res != nullptr;
  if (bf16 && cols % 8 == 0 && cols <= 2 * 256 * 8) {
    const int block = ln_block_for(cols);
    const bool two = cols > (int64_t)block * 8;
    if (has_res) {
      if (two)
        hipLaunchKernelGGL((layer_norm_fwd_bf16_kernel<true, 2>), dim3(grid),
                           dim3(block), 0, stream, out, x, res, sum_out,
                           gamma, beta, mean, rstd, rows, cols, eps);
      else
        hipLaunchKernelGGL((layer_norm_fwd_bf16_kernel<true, 1>), dim3(grid),
                           dim3(block), 0, stream, out, x, res, sum_out,
                           gamma, beta, mean, rstd, rows, cols, eps);
    } else {
      if (two)
        hipLaunchKernelGGL((layer_norm_fwd_bf16_kernel<false, 2>), dim3(grid),
                           dim3(block), 0, stream, out, x, res, sum_out,
                           gamma, beta, mean, rstd, rows, cols, eps);
      else
        hipLaunchKernelGGL((layer_norm_fwd_bf16_kernel<false, 1>), dim3(grid),
                           dim3(block), 0, stream, out, x, res, sum_out,
                           gamma, beta, mean, rstd, rows, cols, eps);
    }
    return;
  }
  if (bf16) {
    if (has_res)
      hipLaunchKernelGGL((layer_norm_fwd_kernel<true, true>), dim3(grid),
                         dim3(kBlock), 0, stream, out, x, res, sum_out,
                         gamma, beta, mean, rstd, rows, cols, eps);
    else
      hipLaunchKernelGGL((layer_norm_fwd_kernel<true, false>), dim3(grid),
                         dim3(kBlock), 0, stream, out, x, res, sum_out,
                         gamma, beta, mean, rstd, rows, cols, eps);
  } else {
    if (has_res)
      hipLaunchKernelGGL((layer_norm_fwd_kernel<false, true>), dim3(grid),
                         dim3(kBlock), 0, stream, out, x, res, sum_out,
                         gamma, beta, mean, rstd, rows, cols, eps);
    else
      hipLaunchKernelGGL((layer_norm_fwd_kernel<false, false>), dim3(grid),
                         dim3(kBlock), 0, stream, out, x, res, sum_out,
                         gamma, beta, mean, rstd, rows, cols, eps);
  }
}

void epl_layer_norm_bwd(void* dx, float* dgamma, float* dbeta, const void* dy,
                        const void* x, const void* gamma, const float* mean,
                        const float* rstd, float* dg_part, float* db_part,
                        int64_t nparts, int64_t rows, int64_t cols,
                        bool bf16, hipStream_t stream) {
  if (bf16 && cols % 8 == 0 && cols <= 2048 && dg_part != nullptr) {
    const int64_t wave_rows = (rows + 3) / 4;
    const int grid = (int)(wave_rows < 1024 ? wave_rows : 1024);
    if (cols <= 512)
      hipLaunchKernelGGL((layer_norm_bwd_bf16_kernel<1, true>), dim3(grid),
                         dim3(256), 0, stream, dx, dg_part, db_part, dy, x,
                         gamma, mean, rstd, rows, cols);
    else if (cols <= 1024)
      hipLaunchKernelGGL((layer_norm_bwd_bf16_kernel<2, true>), dim3(grid),
                         dim3(256), 0, stream, dx, dg_part, db_part, dy, x,
                         gamma, mean, rstd, rows, cols);
    else
      hipLaunchKernelGGL((layer_norm_bwd_bf16_kernel<4, false>),
                         dim3(grid), dim3(256), 0, stream, dx, dg_part,
                         db_part, dy, x, gamma, mean, rstd, rows, cols);
    hipLaunchKernelGGL(ln_bwd_partial_reduce_kernel,
                       dim3((unsigned)((cols + 255) / 256), 64), dim3(256),
                       0, stream, dgamma, dbeta, dg_part, db_part,
                       (int64_t)grid * 4, cols);
    return;
  }
  int grid = (int)(rows < 1024 ? rows : 1024);
  const size_t lds_bytes = (size_t)cols * 2 * sizeof(float);
  if (bf16)
    hipLaunchKernelGGL((layer_norm_bwd_kernel<true>), dim3(grid),
                       dim3(kBlock), lds_bytes, stream, dx, dgamma, dbeta,
                       dy, x, gamma, mean, rstd, rows, cols);
  else
    hipLaunchKernelGGL((layer_norm_bwd_kernel<false>), dim3(grid),
                       dim3(kBlock), lds_bytes, stream, dx, dgamma, dbeta,
                       dy, x, gamma, mean, rstd, rows, cols);
}

void epl_bias_gelu_fwd(void* out, const void* x, const void* bias,
                       int64_t rows, int64_t cols, bool bf16,
                       hipStream_t stream) {
  const int grid = grid_for((rows * cols + 7) / 8);
  if (bf16)
    hipLaunchKernelGGL(bias_gelu_fwd_kernel<true>, dim3(grid), dim3(kBlock), 0,
                       stream, out, x, bias, rows, cols);
  else
    hipLaunchKernelGGL(bias_gelu_fwd_kernel<false>, dim3(grid), dim3(kBlock),
                       0, stream, out, x, bias, rows, cols);
}

void epl_bias_gelu_bwd(void* dx, float* dbias, const void* dy, const void* x,
                       const void* bias, float* db_part, int64_t nwaves,
                       int64_t rows, int64_t cols, bool bf16,
                       hipStream_t stream) {
  if (bf16 && cols % 512 == 0 && db_part != nullptr) {
    const int grid = (int)(nwaves / 4);
    hipLaunchKernelGGL(bias_gelu_bwd_bf16_kernel, dim3(grid), dim3(256), 0,
                       stream, dx, db_part, dy, x, bias, rows, cols);
    const int64_t segs = cols / 512;
    hipLaunchKernelGGL(bias_gelu_db_reduce_kernel,
                       dim3((unsigned)((cols + 255) / 256), 32), dim3(256),
                       0, stream, dbias, db_part, nwaves, segs);
    return;
  }
  int grid = (int)(rows < 1024 ? rows : 1024);
  const size_t lds_bytes = (size_t)cols * sizeof(float);
  if (bf16)
    hipLaunchKernelGGL(bias_gelu_bwd_kernel<true>, dim3(grid), dim3(kBlock),
                       lds_bytes, stream, dx, dbias, dy, x, bias, rows,
                       cols);
  else
    hipLaunchKernelGGL(bias_gelu_bwd_kernel<false>, dim3(grid), dim3(kBlock),
                       lds_bytes, stream, dx, dbias, dy, x, bias, rows,
                       cols);
}

void epl_ce_rowstats(const void* logits, const int64_t* targets,
                     float* row_max, float* row_sumexp, float* target_logit,
                     int64_t rows, int64_t cols, int64_t vocab_begin,
                     int64_t ignore_index, bool bf16, hipStream_t stream) {
  const int grid = (int)(rows < kMaxGrid ? rows : kMaxGrid);
  if (bf16)
    hipLaunchKernelGGL(ce_rowstats_kernel<true>, dim3(grid), dim3(kBlock), 0,
                       stream, logits, targets, row_max, row_sumexp,
                       target_logit, rows, cols, vocab_begin, ignore_index);
  else
    hipLaunchKernelGGL(ce_rowstats_kernel<false>, dim3(grid), dim3(kBlock), 0,
                       stream, logits, targets, row_max, row_sumexp,
                       target_logit, rows, cols, vocab_begin, ignore_index);
}

void epl_ce_backward(void* dlogits, const void* logits, const int64_t* targets,
                     const float* gmax, const float* gsumexp,
                     const float* dloss, int64_t rows, int64_t cols,
                     int64_t vocab_begin, int64_t ignore_index, float scale,
                     bool bf16, hipStream_t stream) {
  const int grid = (int)(rows < kMaxGrid ? rows : kMaxGrid);
  if (bf16)
    hipLaunchKernelGGL(ce_backward_kernel<true>, dim3(grid), dim3(kBlock), 0,
                       stream, dlogits, logits, targets, gmax, gsumexp, dloss,
                       rows, cols, vocab_begin, ignore_index, scale);
  else
    hipLaunchKernelGGL(ce_backward_kernel<false>, dim3(grid), dim3(kBlock), 0,
                       stream, dlogits, logits, targets, gmax, gsumexp, dloss,
                       rows, cols, vocab_begin, ignore_index, scale);
}

void epl_scale(void* p, int64_t n, float s, bool bf16, hipStream_t stream) {
  const int grid = grid_for((n + 7) / 8);
  if (bf16)
    hipLaunchKernelGGL(scale_bf16_kernel, dim3(grid), dim3(kBlock), 0, stream,
                       reinterpret_cast<unsigned short*>(p), n, s);
  else
    hipLaunchKernelGGL(scale_f32_kernel, dim3(grid), dim3(kBlock), 0, stream,
                       reinterpret_cast<float*>(p), n, s);
}

void epl_f32_to_bf16(unsigned short* dst, const float* src, int64_t n,
                     hipStream_t stream) {
  const int grid = grid_for((n + 7) / 8);
  hipLaunchKernelGGL(f32_to_bf16_kernel, dim3(grid), dim3(kBlock), 0, stream,
                     dst, src, n);
}

void epl_bf16_to_f32(float* dst, const unsigned short* src, int64_t n,
                     hipStream_t stream) {
  const int grid = grid_for((n + 7) / 8);
  hipLaunchKernelGGL(bf16_to_f32_kernel, dim3(grid), dim3(kBlock), 0, stream,
                     dst, src, n);
}

void epl_sqnorm(const void* p, int64_t n, float* out, bool bf16,
                hipStream_t stream) {
  const int grid = grid_for(n) < 1024 ? grid_for(n) : 1024;
  if (bf16)
    hipLaunchKernelGGL(sqnorm_kernel<true>, dim3(grid), dim3(kBlock), 0,
                       stream, p, n, out);
  else
    hipLaunchKernelGGL(sqnorm_kernel<false>, dim3(grid), dim3(kBlock), 0,
                       stream, p, n, out);
}


void epl_colsum(void* db, const void* dy, float* partial, int64_t rows,
                int64_t cols, int64_t stripes, bool bf16,
                hipStream_t stream) {
  // size the block to the column count so narrow matrices (cols/8 < 256
  // lanes) don't launch half-idle blocks
  int64_t thr = cols / 8;
  thr = thr < 64 ? 64 : (thr > 256 ? 256 : thr);
  thr = (thr + 63) / 64 * 64;
  dim3 grid((unsigned)((cols / 8 + thr - 1) / thr), (unsigned)stripes);
  if (bf16)
    hipLaunchKernelGGL((colsum_partial_kernel<true>), grid, dim3((unsigned)thr),
                       0, stream, dy, partial, rows, cols, stripes);
  else
    hipLaunchKernelGGL((colsum_partial_kernel<false>), grid,
                       dim3((unsigned)thr), 0, stream, dy, partial, rows,
                       cols, stripes);
  dim3 rgrid((unsigned)((cols + 255) / 256));
  if (bf16)
    hipLaunchKernelGGL((colsum_reduce_kernel<true>), rgrid, dim3(256), 0,
                       stream, db, partial, stripes, cols);
  else
    hipLaunchKernelGGL((colsum_reduce_kernel<false>), rgrid, dim3(256), 0,
                       stream, db, partial, stripes, cols);
}

}  // extern "C"
