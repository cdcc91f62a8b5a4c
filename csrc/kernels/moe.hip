// Fused MoE dispatch/combine kernels (gfx950, bf16, hidden % 8 == 0).
//
// Replaces the torch index machinery of the expert-parallel MLP
// (index_put scatter + advanced-index gather + index_add_ + the
// nonzero-driven boolean select, ~8-10 ms/step on the 12-layer bench —
// op table gpurun_out/r2_moe_opprof.txt) with four deterministic
// row-copy kernels.  Slot assignment stays in torch (stable argsort +
// segmented arange — deterministic, sync-free); the kernels take the
// FULL assignment list and skip over-capacity entries inline, so no
// boolean compaction (and no device->host nonzero sync) ever happens.
//
// Layouts: one 64-lane wave owns one hidden-row copy; rows move as
// bf16x8 (16-byte) vectors, fp32 accumulation where values combine.
// All writes are unique rows -> no atomics, bitwise deterministic.

#include <hip/hip_runtime.h>
#include <cstdint>

using bf16x8 = __attribute__((ext_vector_type(8))) short;

namespace {

__device__ __forceinline__ float mbf2f(short u) {
  union { unsigned int i; float f; } v;
  v.i = ((unsigned int)(unsigned short)u) << 16;
  return v.f;
}

__device__ __forceinline__ unsigned short mf2bf(float f) {
  union { float f; unsigned int i; } v;
  v.f = f;
  unsigned int lsb = (v.i >> 16) & 1u;
  return (unsigned short)((v.i + 0x7fffu + lsb) >> 16);
}

// disp[fe[i], pos[i], :] = x[ft[i], :]   for every kept assignment i
__global__ void moe_dispatch_fwd_kernel(
    short* __restrict__ disp, const short* __restrict__ x,
    const int64_t* __restrict__ fe, const int64_t* __restrict__ pos,
    const int64_t* __restrict__ ft, int64_t m, int64_t hidden,
    int64_t cap) {
  const int64_t waves = ((int64_t)gridDim.x * blockDim.x) >> 6;
  const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const int lane = threadIdx.x & 63;
  for (int64_t i = wid; i < m; i += waves) {
    const int64_t p = pos[i];
    if (p >= cap) continue;
    const short* src = x + ft[i] * hidden;
    short* dst = disp + (fe[i] * cap + p) * hidden;
    for (int64_t c = lane * 8; c < hidden; c += 64 * 8)
      *reinterpret_cast<bf16x8*>(dst + c) =
          *reinterpret_cast<const bf16x8*>(src + c);
  }
}

// dx[t, :] = sum_j ddisp[fe[s], pos[s], :] over t's kept assignments
// (s = inv[t*k + j]); every token has exactly k assignments
__global__ void moe_dispatch_bwd_kernel(
    short* __restrict__ dx, const short* __restrict__ ddisp,
    const int64_t* __restrict__ fe, const int64_t* __restrict__ pos,
    const int64_t* __restrict__ inv, int64_t n_tokens, int64_t k,
    int64_t hidden, int64_t cap) {
  const int64_t waves = ((int64_t)gridDim.x * blockDim.x) >> 6;
  const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const int lane = threadIdx.x & 63;
  for (int64_t t = wid; t < n_tokens; t += waves) {
    for (int64_t c = lane * 8; c < hidden; c += 64 * 8) {
      float acc[8] = {};
      for (int64_t j = 0; j < k; ++j) {
        const int64_t s = inv[t * k + j];
        const int64_t p = pos[s];
        if (p >= cap) continue;
        bf16x8 v = *reinterpret_cast<const bf16x8*>(
            ddisp + (fe[s] * cap + p) * hidden + c);
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[e] += mbf2f(v[e]);
      }
      bf16x8 o;
#pragma unroll
      for (int e = 0; e < 8; ++e) o[e] = (short)mf2bf(acc[e]);
      *reinterpret_cast<bf16x8*>(dx + t * hidden + c) = o;
    }
  }
}

// out[t, :] = sum_j fw[s] * h[fe[s], pos[s], :]  (dropped -> skipped)
__global__ void moe_combine_fwd_kernel(
    short* __restrict__ out, const short* __restrict__ h,
    const float* __restrict__ fw, const int64_t* __restrict__ fe,
    const int64_t* __restrict__ pos, const int64_t* __restrict__ inv,
    int64_t n_tokens, int64_t k, int64_t hidden, int64_t cap) {
  const int64_t waves = ((int64_t)gridDim.x * blockDim.x) >> 6;
  const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const int lane = threadIdx.x & 63;
  for (int64_t t = wid; t < n_tokens; t += waves) {
    for (int64_t c = lane * 8; c < hidden; c += 64 * 8) {
      float acc[8] = {};
      for (int64_t j = 0; j < k; ++j) {
        const int64_t s = inv[t * k + j];
        const int64_t p = pos[s];
        if (p >= cap) continue;
        const float w = fw[s];
        bf16x8 v = *reinterpret_cast<const bf16x8*>(
            h + (fe[s] * cap + p) * hidden + c);
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[e] += w * mbf2f(v[e]);
      }
      bf16x8 o;
#pragma unroll
      for (int e = 0; e < 8; ++e) o[e] = (short)mf2bf(acc[e]);
      *reinterpret_cast<bf16x8*>(out + t * hidden + c) = o;
    }
  }
}

// dh[fe[i], pos[i], :] = fw[i] * dout[ft[i], :]   (unique rows; slots
// never written stay zero — dh must be pre-zeroed by the caller)
// dfw[i] = dot(dout[ft[i]], h[fe[i], pos[i]])     (0 for dropped)
__global__ void moe_combine_bwd_kernel(
    short* __restrict__ dh, float* __restrict__ dfw,
    const short* __restrict__ dout, const short* __restrict__ h,
    const float* __restrict__ fw, const int64_t* __restrict__ fe,
    const int64_t* __restrict__ pos, const int64_t* __restrict__ ft,
    int64_t m, int64_t hidden, int64_t cap) {
  const int64_t waves = ((int64_t)gridDim.x * blockDim.x) >> 6;
  const int64_t wid = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const int lane = threadIdx.x & 63;
  for (int64_t i = wid; i < m; i += waves) {
    const int64_t p = pos[i];
    if (p >= cap) {
      if (lane == 0) dfw[i] = 0.f;
      continue;
    }
    const short* dsrc = dout + ft[i] * hidden;
    const short* hrow = h + (fe[i] * cap + p) * hidden;
    short* drow = dh + (fe[i] * cap + p) * hidden;
    const float w = fw[i];
    float dot = 0.f;
    for (int64_t c = lane * 8; c < hidden; c += 64 * 8) {
      bf16x8 dv = *reinterpret_cast<const bf16x8*>(dsrc + c);
      bf16x8 hv = *reinterpret_cast<const bf16x8*>(hrow + c);
      bf16x8 o;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float d = mbf2f(dv[e]);
        dot += d * mbf2f(hv[e]);
        o[e] = (short)mf2bf(w * d);
      }
      *reinterpret_cast<bf16x8*>(drow + c) = o;
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      dot += __shfl_down(dot, off, 64);
    if (lane == 0) dfw[i] = dot;
  }
}

constexpr int kThreads = 256;

static dim3 row_grid(int64_t rows) {
  // 4 waves per block; >> 256 workgroups fills the 8-XCD chip
  int64_t blocks = (rows + 3) / 4;
  if (blocks > 16384) blocks = 16384;
  if (blocks < 1) blocks = 1;
  return dim3((unsigned)blocks);
}

}  // namespace

extern "C" {

void epl_moe_dispatch_fwd(void* disp, const void* x, const int64_t* fe,
                          const int64_t* pos, const int64_t* ft, int64_t m,
                          int64_t hidden, int64_t cap,
                          hipStream_t stream) {
  hipLaunchKernelGGL(moe_dispatch_fwd_kernel, row_grid(m), dim3(kThreads),
                     0, stream, reinterpret_cast<short*>(disp),
                     reinterpret_cast<const short*>(x), fe, pos, ft, m,
                     hidden, cap);
}

void epl_moe_dispatch_bwd(void* dx, const void* ddisp, const int64_t* fe,
                          const int64_t* pos, const int64_t* inv,
                          int64_t n_tokens, int64_t k, int64_t hidden,
                          int64_t cap, hipStream_t stream) {
  hipLaunchKernelGGL(moe_dispatch_bwd_kernel, row_grid(n_tokens),
                     dim3(kThreads), 0, stream,
                     reinterpret_cast<short*>(dx),
                     reinterpret_cast<const short*>(ddisp), fe, pos, inv,
                     n_tokens, k, hidden, cap);
}

void epl_moe_combine_fwd(void* out, const void* h, const float* fw,
                         const int64_t* fe, const int64_t* pos,
                         const int64_t* inv, int64_t n_tokens, int64_t k,
                         int64_t hidden, int64_t cap, hipStream_t stream) {
  hipLaunchKernelGGL(moe_combine_fwd_kernel, row_grid(n_tokens),
                     dim3(kThreads), 0, stream,
                     reinterpret_cast<short*>(out),
                     reinterpret_cast<const short*>(h), fw, fe, pos, inv,
                     n_tokens, k, hidden, cap);
}

void epl_moe_combine_bwd(void* dh, float* dfw, const void* dout,
                         const void* h, const float* fw, const int64_t* fe,
                         const int64_t* pos, const int64_t* ft, int64_t m,
                         int64_t hidden, int64_t cap, hipStream_t stream) {
  hipLaunchKernelGGL(moe_combine_bwd_kernel, row_grid(m), dim3(kThreads),
                     0, stream, reinterpret_cast<short*>(dh), dfw,
                     reinterpret_cast<const short*>(dout),
                     reinterpret_cast<const short*>(h), fw, fe, pos, ft, m,
                     hidden, cap);
  }

}  // extern "C"
