// CDNA4 (gfx950 / MI355X) kernels for sheeprl-amd.
//
// Replaces the PyTorch-op subgraphs listed in SURVEY.md §2.8 with fused HIP
// kernels: symlog/symexp (utils.py:148-154), fused LayerNorm+SiLU (the MLP/CNN
// epilogue, models.py:16/122), the LayerNormGRUCell post-GEMM gate math
// (models.py:396-403), the GAE and λ-return reverse scans (utils.py:63-100,
// dreamer_v3/utils.py:66-77), multi-tensor Adam/EMA (optim) and uint8 obs
// normalization (dreamer_v3/utils.py:80-91).
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//  * wave = 64 lanes; blocks are multiples of 64 (256 default).
//  * reductions: __shfl_xor over 64-lane waves, then LDS across waves.
//  * elementwise kernels are grid-stride, vectorized 16B/lane where the
//    tensor is contiguous and size-aligned (G13).
//  * all accumulation in fp32 regardless of storage dtype (bf16-true safe).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <vector>

#define CHECK_IN(x) TORCH_CHECK(x.is_cuda() && x.is_contiguous(), #x " must be contiguous CUDA tensor")

namespace {

constexpr int kBlock = 256;

// fast transcendental variants for the bf16/half kernel instantiations:
// v_exp_f32 (~2 instructions) vs libm expf (~16); relative error ~1e-6 is
// far below bf16 resolution.  fp32 instantiations keep libm for the exact
// numerics tests.
template <typename T>
__device__ __forceinline__ float texp(float v) {
  if constexpr (sizeof(T) == 2)
    return __expf(v);
  else
    return expf(v);
}
template <typename T>
__device__ __forceinline__ float tlog(float v) {
  if constexpr (sizeof(T) == 2)
    return __logf(v);
  else
    return logf(v);
}
template <typename T>
__device__ __forceinline__ float ttanh(float v) {
  if constexpr (sizeof(T) == 2)
    return 1.f - 2.f / (__expf(2.f * v) + 1.f);
  else
    return tanhf(v);
}

__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// block-wide sum over kBlock threads (4 waves)
__device__ __forceinline__ float block_sum(float v, float* lds) {
  v = wave_sum(v);
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  float out = 0.f;
  if (threadIdx.x < (blockDim.x >> 6)) out = lds[threadIdx.x];
  out = wave_sum(out);  // lanes beyond nwaves hold 0
  // broadcast via lds
  if (threadIdx.x == 0) lds[8] = out;
  __syncthreads();
  return lds[8];
}

// block-wide sum of TWO accumulators at once (one barrier pair instead of
// two full block_sum rounds); lds must hold >= 18 floats
__device__ __forceinline__ void block_sum2(float& a, float& b, float* lds) {
  a = wave_sum(a);
  b = wave_sum(b);
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  if (lane == 0) {
    lds[wid] = a;
    lds[8 + wid] = b;
  }
  __syncthreads();
  float oa = 0.f, ob = 0.f;
  if (threadIdx.x < (blockDim.x >> 6)) {
    oa = lds[threadIdx.x];
    ob = lds[8 + threadIdx.x];
  }
  oa = wave_sum(oa);
  ob = wave_sum(ob);
  if (threadIdx.x == 0) {
    lds[16] = oa;
    lds[17] = ob;
  }
  __syncthreads();
  a = lds[16];
  b = lds[17];
}

template <typename T>
__device__ __forceinline__ float ld(const T* p, long i) {
  return static_cast<float>(p[i]);
}
template <>
__device__ __forceinline__ float ld<__hip_bfloat16>(const __hip_bfloat16* p, long i) {
  return __bfloat162float(p[i]);
}

template <typename T>
__device__ __forceinline__ void st(T* p, long i, float v) {
  p[i] = static_cast<T>(v);
}
template <>
__device__ __forceinline__ void st<__hip_bfloat16>(__hip_bfloat16* p, long i, float v) {
  p[i] = __float2bfloat16(v);
}

// ---------------------------------------------------------------------------
// symlog / symexp
// ---------------------------------------------------------------------------

template <typename T, int OP>
__global__ void symmath_kernel(const T* __restrict__ x, const T* __restrict__ gy, T* __restrict__ y, long n) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n; i += (long)gridDim.x * blockDim.x) {
    float v = ld(x, i);
    float a = fabsf(v);
    float s = v >= 0.f ? 1.f : -1.f;
    float out;
    if (OP == 0) out = s * log1pf(a);                         // symlog fwd
    else if (OP == 1) out = ld(gy, i) / (1.f + a);            // symlog bwd
    else if (OP == 2) out = s * (expf(a) - 1.f);              // symexp fwd
    else out = ld(gy, i) * expf(a);                           // symexp bwd
    st(y, i, out);
  }
}

template <int OP>
torch::Tensor symmath(const torch::Tensor& x, const c10::optional<torch::Tensor>& gy) {
  CHECK_IN(x);
  auto y = torch::empty_like(x);
  long n = x.numel();
  if (n == 0) return y;
  int blocks = (int)std::min((n + kBlock - 1) / kBlock, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "symmath", [&] {
    using T = scalar_t;
    const T* gp = gy.has_value() ? (const T*)gy->data_ptr() : nullptr;
    hipLaunchKernelGGL((symmath_kernel<T, OP>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       (const T*)x.data_ptr(), gp, (T*)y.data_ptr(), n);
  });
  return y;
}

// ---------------------------------------------------------------------------
// fused LayerNorm (+SiLU) forward/backward; rows of length D
// ---------------------------------------------------------------------------

// W can be a different float dtype from X (bf16-true modules keep bf16 affine
// params; the GPU tests pass fp32) — templated on both to avoid per-call
// conversion launches.
template <typename T, typename TW, bool SILU>
__global__ void ln_act_fwd_kernel(const T* __restrict__ x, const TW* __restrict__ w,
                                  const TW* __restrict__ b, T* __restrict__ y, float* __restrict__ mean_out,
                                  float* __restrict__ rstd_out, int D, float eps, long ys, bool cached) {
  __shared__ float lds[18];
  const long row = blockIdx.x;
  const T* xr = x + row * (long)D;
  T* yr = y + row * ys;
  // register-cached single-read path: with the scan's tiny row counts the
  // kernel is latency-bound, so one global pass + one fused sum/sumsq
  // reduction (instead of mean pass, barrier, var pass, barrier) is ~1/3
  // fewer round trips; 16 floats/thread covers D <= 4096 at 256 threads.
  // The cache array lives in scratch (runtime-bounded indexing), which is
  // invisible at <=64 rows but dominates at behaviour-batch row counts —
  // the host gates `cached` by N (measured: 16384x1024 fwd 48.9 us cached).
  if (cached && D <= 16 * (int)blockDim.x) {
    float cache[16];
    float s = 0.f, s2 = 0.f;
    int cnt = 0;
    for (int j = threadIdx.x; j < D; j += blockDim.x) {
      float v = ld(xr, j);
      cache[cnt++] = v;
      s += v;
      s2 += v * v;
    }
    block_sum2(s, s2, lds);
    float mean = s / D;
    float var = s2 / D - mean * mean;
    float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
    if (threadIdx.x == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    cnt = 0;
    for (int j = threadIdx.x; j < D; j += blockDim.x) {
      float xhat = (cache[cnt++] - mean) * rstd;
      float z = xhat * ld(w, j) + ld(b, j);
      if (SILU) z = z / (1.f + texp<T>(-z));
      st(yr, j, z);
    }
    return;
  }
  float s = 0.f;
  for (int j = threadIdx.x; j < D; j += blockDim.x) s += ld(xr, j);
  float mean = block_sum(s, lds) / D;
  __syncthreads();
  float s2 = 0.f;
  for (int j = threadIdx.x; j < D; j += blockDim.x) {
    float d = ld(xr, j) - mean;
    s2 += d * d;
  }
  float var = block_sum(s2, lds) / D;
  float rstd = rsqrtf(var + eps);
  if (threadIdx.x == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  for (int j = threadIdx.x; j < D; j += blockDim.x) {
    float xhat = (ld(xr, j) - mean) * rstd;
    float z = xhat * ld(w, j) + ld(b, j);
    if (SILU) z = z / (1.f + texp<T>(-z));
    st(yr, j, z);
  }
}

// Wave-per-row variant for short rows (conv-channel LayerNorms: D <= 256 but
// N up to ~1M rows).  4 waves per block, grid-stride over rows.
template <typename T, typename TW, bool SILU>
__global__ void ln_act_fwd_small_kernel(const T* __restrict__ x, const TW* __restrict__ w,
                                        const TW* __restrict__ b, T* __restrict__ y,
                                        float* __restrict__ mean_out, float* __restrict__ rstd_out, long N, int D,
                                        float eps, long ys) {
  const int lane = threadIdx.x & 63;
  const long wave = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const long nwaves = (long)gridDim.x * (blockDim.x >> 6);
  for (long row = wave; row < N; row += nwaves) {
    const T* xr = x + row * (long)D;
    T* yr = y + row * ys;
    float s = 0.f, s2 = 0.f;
    for (int j = lane; j < D; j += 64) {
      float v = ld(xr, j);
      s += v;
      s2 += v * v;
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      s += __shfl_xor(s, off, 64);
      s2 += __shfl_xor(s2, off, 64);
    }
    float mean = s / D;
    float var = s2 / D - mean * mean;
    float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
    if (lane == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    for (int j = lane; j < D; j += 64) {
      float xhat = (ld(xr, j) - mean) * rstd;
      float z = xhat * ld(w, j) + ld(b, j);
      if (SILU) z = z / (1.f + texp<T>(-z));
      st(yr, j, z);
    }
  }
}

// Vectorized wave-segmented LN kernels for channels-last conv rows: each lane
// loads a 16-byte vector (V elems), L = D/V lanes form one row, a 64-lane
// wave handles 64/L rows per iteration with fully coalesced 1-KB accesses.
// The scalar wave-per-row small kernel reached only ~0.4 TB/s on [1M, 32]
// rows (64 B per wave-load); this layout is bandwidth-bound.
template <typename T, int V>
union LnVec {
  uint4 u;
  T e[V];
};

template <typename T, typename TW, bool SILU, int L>
__global__ void __launch_bounds__(kBlock) ln_act_fwd_cl_kernel(const T* __restrict__ x, const TW* __restrict__ w,
                                     const TW* __restrict__ b, T* __restrict__ y,
                                     float* __restrict__ mean_out, float* __restrict__ rstd_out, long R, int D,
                                     float eps) {
  constexpr int V = 16 / sizeof(T);
  constexpr int RPW = 64 / L;  // floor for non-pow2 L (12/24: conv widths
                               // 96/192 bf16); lanes past RPW*L idle
  constexpr int UNROLL = 4;  // row-blocks in flight per wave: one 16-B load
                             // per lane is latency-bound at ~3 TB/s
  const int lane = threadIdx.x & 63;
  const int lig = lane % L;
  const int grp = lane / L;
  const bool lane_ok = grp < RPW;
  const long wave = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const long nwaves = (long)gridDim.x * (blockDim.x >> 6);
  float wv[V], bv[V];
#pragma unroll
  for (int e = 0; e < V; ++e) {
    wv[e] = ld(w, lig * V + e);
    bv[e] = ld(b, lig * V + e);
  }
  const long stride = nwaves * RPW;
  for (long rb = wave * RPW; rb < R; rb += stride * UNROLL) {
    LnVec<T, V> xv[UNROLL];
    long rows[UNROLL];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      rows[u] = rb + u * stride + grp;
      if (lane_ok && rows[u] < R) xv[u].u = *reinterpret_cast<const uint4*>(x + rows[u] * (long)D + lig * V);
    }
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      const long row = rows[u];
      const bool active = lane_ok && row < R;
      float s = 0.f, s2 = 0.f;
      if (active) {
#pragma unroll
        for (int e = 0; e < V; ++e) {
          float v = ld(xv[u].e, e);
          s += v;
          s2 += v * v;
        }
      }
      if constexpr ((L & (L - 1)) == 0) {
#pragma unroll
        for (int off = 1; off < L; off <<= 1) {
          s += __shfl_xor(s, off, 64);
          s2 += __shfl_xor(s2, off, 64);
        }
      } else {
        // non-pow2 group: gather-loop sum over the group's L lanes
        const int base = grp * L;
        float a = s, a2 = s2;
        s = 0.f;
        s2 = 0.f;
#pragma unroll
        for (int i = 0; i < L; ++i) {
          s += __shfl(a, (base + i) & 63, 64);
          s2 += __shfl(a2, (base + i) & 63, 64);
        }
      }
      if (active) {
        float mean = s / D;
        float var = s2 / D - mean * mean;
        float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
        if (lig == 0) {
          mean_out[row] = mean;
          rstd_out[row] = rstd;
        }
        LnVec<T, V> yv;
#pragma unroll
        for (int e = 0; e < V; ++e) {
          float z = (ld(xv[u].e, e) - mean) * rstd * wv[e] + bv[e];
          if (SILU) z = z / (1.f + expf(-z));
          st(yv.e, e, z);
        }
        *reinterpret_cast<uint4*>(y + row * (long)D + lig * V) = yv.u;
      }
    }
  }
}

template <typename T, typename TW, bool SILU, int L>
__global__ void __launch_bounds__(kBlock) ln_act_bwd_cl_kernel(const T* __restrict__ gy, const T* __restrict__ x,
                                     const TW* __restrict__ w, const TW* __restrict__ b,
                                     const float* __restrict__ mean, const float* __restrict__ rstd,
                                     T* __restrict__ gx, float* __restrict__ gw, float* __restrict__ gb, long R,
                                     int D) {
  constexpr int V = 16 / sizeof(T);
  constexpr int RPW = 64 / L;  // floor for non-pow2 L; lanes past RPW*L idle
  extern __shared__ __attribute__((aligned(16))) float smem[];  // [2*D]
  float* gw_s = smem;
  float* gb_s = smem + D;
  for (int j = threadIdx.x; j < 2 * D; j += blockDim.x) smem[j] = 0.f;
  __syncthreads();
  const int lane = threadIdx.x & 63;
  const int lig = lane % L;
  const int grp = lane / L;
  const bool lane_ok = grp < RPW;
  const long wave = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const long nwaves = (long)gridDim.x * (blockDim.x >> 6);
  float wv[V], bv[V], gwa[V], gba[V];
#pragma unroll
  for (int e = 0; e < V; ++e) {
    wv[e] = ld(w, lig * V + e);
    bv[e] = ld(b, lig * V + e);
    gwa[e] = 0.f;
    gba[e] = 0.f;
  }
  const long stride = nwaves * RPW;
  constexpr int UNROLL = 2;  // 2 row-blocks x 2 tensors = 4 loads in flight
  for (long rb = wave * RPW; rb < R; rb += stride * UNROLL) {
    LnVec<T, V> xv[UNROLL], gv[UNROLL];
    long rows[UNROLL];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      rows[u] = rb + u * stride + grp;
      if (lane_ok && rows[u] < R) {
        xv[u].u = *reinterpret_cast<const uint4*>(x + rows[u] * (long)D + lig * V);
        gv[u].u = *reinterpret_cast<const uint4*>(gy + rows[u] * (long)D + lig * V);
      }
    }
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      const long row = rows[u];
      const bool active = lane_ok && row < R;
      float xh[V], gz[V];
      float s1 = 0.f, s2 = 0.f;
      float rs = 0.f;
      if (active) {
        const float m = mean[row];
        rs = rstd[row];
#pragma unroll
        for (int e = 0; e < V; ++e) {
          xh[e] = (ld(xv[u].e, e) - m) * rs;
          float g = ld(gv[u].e, e);
          if (SILU) {
            float z = xh[e] * wv[e] + bv[e];
            float sig = 1.f / (1.f + expf(-z));
            g *= sig * (1.f + z * (1.f - sig));
          }
          gz[e] = g;
          gwa[e] += g * xh[e];
          gba[e] += g;
          float gxhat = g * wv[e];
          s1 += gxhat;
          s2 += gxhat * xh[e];
        }
      }
      if constexpr ((L & (L - 1)) == 0) {
#pragma unroll
        for (int off = 1; off < L; off <<= 1) {
          s1 += __shfl_xor(s1, off, 64);
          s2 += __shfl_xor(s2, off, 64);
        }
      } else {
        const int base = grp * L;
        float a1 = s1, a2 = s2;
        s1 = 0.f;
        s2 = 0.f;
#pragma unroll
        for (int i = 0; i < L; ++i) {
          s1 += __shfl(a1, (base + i) & 63, 64);
          s2 += __shfl(a2, (base + i) & 63, 64);
        }
      }
      if (active) {
        const float S1 = s1 / D, S2 = s2 / D;
        LnVec<T, V> ov;
#pragma unroll
        for (int e = 0; e < V; ++e) st(ov.e, e, (gz[e] * wv[e] - S1 - xh[e] * S2) * rs);
        *reinterpret_cast<uint4*>(gx + row * (long)D + lig * V) = ov.u;
      }
    }
  }
  // fold groups within the wave (lanes sharing lane%L hold the same columns)
  if constexpr ((L & (L - 1)) == 0) {
#pragma unroll
    for (int off = L; off < 64; off <<= 1) {
#pragma unroll
      for (int e = 0; e < V; ++e) {
        gwa[e] += __shfl_xor(gwa[e], off, 64);
        gba[e] += __shfl_xor(gba[e], off, 64);
      }
    }
  } else {
#pragma unroll
    for (int e = 0; e < V; ++e) {
      float aw = gwa[e], ab = gba[e];
      gwa[e] = 0.f;
      gba[e] = 0.f;
#pragma unroll
      for (int g = 0; g < RPW; ++g) {
        gwa[e] += __shfl(aw, (lig + g * L) & 63, 64);
        gba[e] += __shfl(ab, (lig + g * L) & 63, 64);
      }
    }
  }
  if (grp == 0) {
#pragma unroll
    for (int e = 0; e < V; ++e) {
      atomicAdd(&gw_s[lig * V + e], gwa[e]);
      atomicAdd(&gb_s[lig * V + e], gba[e]);
    }
  }
  __syncthreads();
  for (int j = threadIdx.x; j < D; j += blockDim.x) {
    atomicAdd(&gw[j], gw_s[j]);
    atomicAdd(&gb[j], gb_s[j]);
  }
}

// Vectorized wave-per-row LN kernels for the behaviour-MLP shapes (many
// rows x 512 < D <= 2048): each lane holds K 16-B vectors of its row, so a
// row needs nlanes = D/(K*V) <= 64 lanes and the LN statistics reduce with
// wave shuffles ONLY — no LDS, no barriers in the row loop (a first
// block-per-row version paid a block_sum2 barrier round per row and ran
// 127 us on [16384, 1024]).  gw/gb accumulate in registers over the row
// loop (each lane owns fixed columns), fold across the block's waves in
// LDS once, and flush with one atomicAdd per column per block — blocks
// bounded so the flush stays off the critical path.
template <typename T, typename TW, bool SILU, int K>
__global__ void __launch_bounds__(kBlock) ln_act_fwd_v_kernel(const T* __restrict__ x, const TW* __restrict__ w,
                                    const TW* __restrict__ b, T* __restrict__ y,
                                    float* __restrict__ mean_out, float* __restrict__ rstd_out, long N,
                                    int D, float eps, long ys) {
  constexpr int V = 16 / sizeof(T);
  const int lane = threadIdx.x & 63;
  const int nlanes = D / (K * V);
  const bool own = lane < nlanes;
  const long wave = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const long nwaves = (long)gridDim.x * (blockDim.x >> 6);
  float wv[K * V], bv[K * V];
  if (own) {
#pragma unroll
    for (int k = 0; k < K; ++k)
#pragma unroll
      for (int e = 0; e < V; ++e) {
        wv[k * V + e] = ld(w, (k * nlanes + lane) * V + e);
        bv[k * V + e] = ld(b, (k * nlanes + lane) * V + e);
      }
  }
  LnVec<T, V> xv[K], xn[K];
  if (own && wave < N) {
#pragma unroll
    for (int k = 0; k < K; ++k)
      xv[k].u = *reinterpret_cast<const uint4*>(x + wave * (long)D + (k * nlanes + lane) * V);
  }
  for (long row = wave; row < N; row += nwaves) {
    T* yr = y + row * ys;
    // issue the NEXT row's loads before this row's math so the row loop
    // keeps >1 memory latency in flight per wave (no barriers to hide it)
    const long nrow = row + nwaves;
    if (own && nrow < N) {
#pragma unroll
      for (int k = 0; k < K; ++k)
        xn[k].u = *reinterpret_cast<const uint4*>(x + nrow * (long)D + (k * nlanes + lane) * V);
    }
    float s = 0.f, s2 = 0.f;
    if (own) {
#pragma unroll
      for (int k = 0; k < K; ++k) {
#pragma unroll
        for (int e = 0; e < V; ++e) {
          float v = ld(xv[k].e, e);
          s += v;
          s2 += v * v;
        }
      }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      s += __shfl_xor(s, off, 64);
      s2 += __shfl_xor(s2, off, 64);
    }
    const float mean = s / D;
    const float var = s2 / D - mean * mean;
    const float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
    if (lane == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    if (own) {
#pragma unroll
      for (int k = 0; k < K; ++k) {
        LnVec<T, V> yv;
#pragma unroll
        for (int e = 0; e < V; ++e) {
          float z = (ld(xv[k].e, e) - mean) * rstd * wv[k * V + e] + bv[k * V + e];
          if (SILU) z = z / (1.f + texp<T>(-z));
          st(yv.e, e, z);
        }
        *reinterpret_cast<uint4*>(yr + (k * nlanes + lane) * V) = yv.u;
      }
    }
#pragma unroll
    for (int k = 0; k < K; ++k) xv[k] = xn[k];
  }
}

template <typename T, typename TW, bool SILU, int K>
__global__ void __launch_bounds__(kBlock) ln_act_bwd_v_kernel(const T* __restrict__ gy, const T* __restrict__ x,
                                    const TW* __restrict__ w, const TW* __restrict__ b,
                                    const float* __restrict__ mean, const float* __restrict__ rstd,
                                    T* __restrict__ gx, float* __restrict__ gw, float* __restrict__ gb,
                                    long N, int D, long gys) {
  constexpr int V = 16 / sizeof(T);
  extern __shared__ __attribute__((aligned(16))) float smem[];  // [2*D]
  float* gw_s = smem;
  float* gb_s = smem + D;
  for (int j = threadIdx.x; j < 2 * D; j += blockDim.x) smem[j] = 0.f;
  __syncthreads();
  const int lane = threadIdx.x & 63;
  const int nlanes = D / (K * V);
  const bool own = lane < nlanes;
  const long wave = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const long nwaves = (long)gridDim.x * (blockDim.x >> 6);
  float wv[K * V], bv[K * V], gwa[K * V], gba[K * V];
#pragma unroll
  for (int k = 0; k < K * V; ++k) {
    gwa[k] = 0.f;
    gba[k] = 0.f;
  }
  if (own) {
#pragma unroll
    for (int k = 0; k < K; ++k)
#pragma unroll
      for (int e = 0; e < V; ++e) {
        wv[k * V + e] = ld(w, (k * nlanes + lane) * V + e);
        bv[k * V + e] = ld(b, (k * nlanes + lane) * V + e);
      }
  }
  LnVec<T, V> xv[K], gv[K], xn[K], gn[K];
  if (own && wave < N) {
#pragma unroll
    for (int k = 0; k < K; ++k) {
      xv[k].u = *reinterpret_cast<const uint4*>(x + wave * (long)D + (k * nlanes + lane) * V);
      gv[k].u = *reinterpret_cast<const uint4*>(gy + wave * gys + (k * nlanes + lane) * V);
    }
  }
  for (long row = wave; row < N; row += nwaves) {
    T* gxr = gx + row * (long)D;
    const long nrow = row + nwaves;
    if (own && nrow < N) {
#pragma unroll
      for (int k = 0; k < K; ++k) {
        xn[k].u = *reinterpret_cast<const uint4*>(x + nrow * (long)D + (k * nlanes + lane) * V);
        gn[k].u = *reinterpret_cast<const uint4*>(gy + nrow * gys + (k * nlanes + lane) * V);
      }
    }
    const float m = mean[row], rs = rstd[row];
    // gz (post-SILU) is cached in registers across the two passes — the PMC
    // counters showed this kernel 50% active-issue (VALU-bound), so the
    // SILU-derivative recompute was the cost, not the loads; with
    // __launch_bounds__(256) the 16 extra VGPRs no longer spill.  xh is
    // recomputed (2 ops).
    float cgz[K * V];
    float s1 = 0.f, s2 = 0.f;
    if (own) {
#pragma unroll
      for (int k = 0; k < K; ++k) {
#pragma unroll
        for (int e = 0; e < V; ++e) {
          const int c = k * V + e;
          const float xh = (ld(xv[k].e, e) - m) * rs;
          float g = ld(gv[k].e, e);
          if (SILU) {
            float z = xh * wv[c] + bv[c];
            float sig = 1.f / (1.f + texp<T>(-z));
            g *= sig * (1.f + z * (1.f - sig));
          }
          cgz[c] = g;
          float gxhat = g * wv[c];
          s1 += gxhat;
          s2 += gxhat * xh;
        }
      }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      s1 += __shfl_xor(s1, off, 64);
      s2 += __shfl_xor(s2, off, 64);
    }
    if (own) {
      const float S1 = s1 / D, S2 = s2 / D;
#pragma unroll
      for (int k = 0; k < K; ++k) {
        LnVec<T, V> ov;
#pragma unroll
        for (int e = 0; e < V; ++e) {
          const int c = k * V + e;
          const float xh = (ld(xv[k].e, e) - m) * rs;
          const float g = cgz[c];
          gwa[c] += g * xh;
          gba[c] += g;
          st(ov.e, e, (g * wv[c] - S1 - xh * S2) * rs);
        }
        *reinterpret_cast<uint4*>(gxr + (k * nlanes + lane) * V) = ov.u;
      }
    }
#pragma unroll
    for (int k = 0; k < K; ++k) {
      xv[k] = xn[k];
      gv[k] = gn[k];
    }
  }
  // fold the block's waves through LDS (all waves own the same columns)
  if (own) {
#pragma unroll
    for (int k = 0; k < K; ++k)
#pragma unroll
      for (int e = 0; e < V; ++e) {
        const int j = (k * nlanes + lane) * V + e;
        atomicAdd(&gw_s[j], gwa[k * V + e]);
        atomicAdd(&gb_s[j], gba[k * V + e]);
      }
  }
  __syncthreads();
  for (int j = threadIdx.x; j < D; j += blockDim.x) {
    atomicAdd(&gw[j], gw_s[j]);
    atomicAdd(&gb[j], gb_s[j]);
  }
}

// returns L (lanes per row) when the vectorized channels-last path applies
template <typename T>
int ln_cl_lanes(int D, long stride) {
  constexpr int V = 16 / sizeof(T);
  if (stride != D || D % V != 0) return 0;
  int L = D / V;
  if (L < 2 || L > 64) return 0;
  if ((L & (L - 1)) == 0) return L;
  // non-pow2 lane groups supported for the conv channel widths (96/192
  // bf16 -> L=12/24): floor(64/L) rows per wave, tail lanes idle
  if (L == 12 || L == 24) return L;
  return 0;
}

template <typename T, typename TW, bool SILU>
void launch_ln_fwd_cl(int L, long R, hipStream_t st, const T* x, const TW* w, const TW* b, T* y, float* mo,
                      float* ro, int D, float eps) {
  int rpb = (kBlock / 64) * (64 / L);
  int blocks = (int)std::min((R + rpb - 1) / rpb, (long)2048);
#define SHEEP_LN_FWD_CASE(LV)                                                                              \
  case LV:                                                                                                 \
    hipLaunchKernelGGL((ln_act_fwd_cl_kernel<T, TW, SILU, LV>), dim3(blocks), dim3(kBlock), 0, st, x, w, b, \
                       y, mo, ro, R, D, eps);                                                              \
    break;
  switch (L) {
    SHEEP_LN_FWD_CASE(2)
    SHEEP_LN_FWD_CASE(4)
    SHEEP_LN_FWD_CASE(8)
    SHEEP_LN_FWD_CASE(12)
    SHEEP_LN_FWD_CASE(16)
    SHEEP_LN_FWD_CASE(24)
    SHEEP_LN_FWD_CASE(32)
    SHEEP_LN_FWD_CASE(64)
  }
#undef SHEEP_LN_FWD_CASE
}

template <typename T, typename TW, bool SILU>
void launch_ln_bwd_cl(int L, long R, hipStream_t st, const T* gy, const T* x, const TW* w, const TW* b,
                      const float* mean, const float* rstd, T* gx, float* gw, float* gb, int D) {
  int rpb = (kBlock / 64) * (64 / L);
  int blocks = (int)std::min((R + rpb - 1) / rpb, (long)2048);
  // each block flushes 2*D atomicAdds at the end; for mid-sized N that flush
  // dominates — keep >=32 rows of real work per block.  The parallelism
  // floor is tunable for measurement (SHEEPRL_AMD_LN_BWD_FLOOR, default 64).
  static const long kFloor = [] {
    const char* s = getenv("SHEEPRL_AMD_LN_BWD_FLOOR");
    return s ? atol(s) : 64L;
  }();
  blocks = (int)std::min((long)blocks, std::max(kFloor, (R + 31) / 32));
  size_t shmem = 2 * (size_t)D * sizeof(float);
#define SHEEP_LN_BWD_CASE(LV)                                                                               \
  case LV:                                                                                                  \
    hipLaunchKernelGGL((ln_act_bwd_cl_kernel<T, TW, SILU, LV>), dim3(blocks), dim3(kBlock), shmem, st, gy,  \
                       x, w, b, mean, rstd, gx, gw, gb, R, D);                                              \
    break;
  switch (L) {
    SHEEP_LN_BWD_CASE(2)
    SHEEP_LN_BWD_CASE(4)
    SHEEP_LN_BWD_CASE(8)
    SHEEP_LN_BWD_CASE(12)
    SHEEP_LN_BWD_CASE(16)
    SHEEP_LN_BWD_CASE(24)
    SHEEP_LN_BWD_CASE(32)
    SHEEP_LN_BWD_CASE(64)
  }
#undef SHEEP_LN_BWD_CASE
}

void ln_act_fwd_core(const torch::Tensor& x, const torch::Tensor& w, const torch::Tensor& b, double eps,
                     bool silu, torch::Tensor& y, torch::Tensor& mean, torch::Tensor& rstd, long ys) {
  long N = x.size(0);
  int D = (int)x.size(1);
  auto wc = w.contiguous();
  auto bc = b.contiguous();
  auto stream = at::cuda::getCurrentCUDAStream();
  const bool small = D <= 256;
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "ln_act_fwd", [&] {
    using T = scalar_t;
    AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, wc.scalar_type(), "ln_act_fwd_w", [&] {
      using TW = scalar_t;
      const int L = ln_cl_lanes<T>(D, ys);
      if (L) {
        if (silu)
          launch_ln_fwd_cl<T, TW, true>(L, N, stream.stream(), (const T*)x.data_ptr(), (const TW*)wc.data_ptr(),
                                        (const TW*)bc.data_ptr(), (T*)y.data_ptr(), mean.data_ptr<float>(),
                                        rstd.data_ptr<float>(), D, (float)eps);
        else
          launch_ln_fwd_cl<T, TW, false>(L, N, stream.stream(), (const T*)x.data_ptr(), (const TW*)wc.data_ptr(),
                                         (const TW*)bc.data_ptr(), (T*)y.data_ptr(), mean.data_ptr<float>(),
                                         rstd.data_ptr<float>(), D, (float)eps);
      } else if (int K = (N > 64 && ys % (16 / (int)sizeof(T)) == 0)
                              ? ((D % (2 * (16 / (int)sizeof(T))) == 0 && D / (2 * (16 / (int)sizeof(T))) <= 64)
                                     ? 2
                                     : ((D % (4 * (16 / (int)sizeof(T))) == 0 &&
                                         D / (4 * (16 / (int)sizeof(T))) <= 64)
                                            ? 4
                                            : 0))
                              : 0) {
        // vectorized wave-per-row (behaviour-MLP shapes): no barriers in
        // the row loop, wave-shuffle LN stats, 16-B lane loads
        int blocks = (int)std::min((N + 3) / 4, (long)2048);
        if (silu) {
          if (K == 2)
            hipLaunchKernelGGL((ln_act_fwd_v_kernel<T, TW, true, 2>), dim3(blocks), dim3(kBlock), 0,
                               stream.stream(), (const T*)x.data_ptr(), (const TW*)wc.data_ptr(),
                               (const TW*)bc.data_ptr(), (T*)y.data_ptr(), mean.data_ptr<float>(),
                               rstd.data_ptr<float>(), N, D, (float)eps, ys);
          else
            hipLaunchKernelGGL((ln_act_fwd_v_kernel<T, TW, true, 4>), dim3(blocks), dim3(kBlock), 0,
                               stream.stream(), (const T*)x.data_ptr(), (const TW*)wc.data_ptr(),
                               (const TW*)bc.data_ptr(), (T*)y.data_ptr(), mean.data_ptr<float>(),
                               rstd.data_ptr<float>(), N, D, (float)eps, ys);
        } else {
          if (K == 2)
            hipLaunchKernelGGL((ln_act_fwd_v_kernel<T, TW, false, 2>), dim3(blocks), dim3(kBlock), 0,
                               stream.stream(), (const T*)x.data_ptr(), (const TW*)wc.data_ptr(),
                               (const TW*)bc.data_ptr(), (T*)y.data_ptr(), mean.data_ptr<float>(),
                               rstd.data_ptr<float>(), N, D, (float)eps, ys);
          else
            hipLaunchKernelGGL((ln_act_fwd_v_kernel<T, TW, false, 4>), dim3(blocks), dim3(kBlock), 0,
                               stream.stream(), (const T*)x.data_ptr(), (const TW*)wc.data_ptr(),
                               (const TW*)bc.data_ptr(), (T*)y.data_ptr(), mean.data_ptr<float>(),
                               rstd.data_ptr<float>(), N, D, (float)eps, ys);
        }
      } else if (small) {
        int blocks = (int)std::min((N + 3) / 4, (long)2048);
        if (silu)
          hipLaunchKernelGGL((ln_act_fwd_small_kernel<T, TW, true>), dim3(blocks), dim3(kBlock), 0,
                             stream.stream(), (const T*)x.data_ptr(), (const TW*)wc.data_ptr(),
                             (const TW*)bc.data_ptr(), (T*)y.data_ptr(), mean.data_ptr<float>(),
                             rstd.data_ptr<float>(), N, D, (float)eps, ys);
        else
          hipLaunchKernelGGL((ln_act_fwd_small_kernel<T, TW, false>), dim3(blocks), dim3(kBlock), 0,
                             stream.stream(), (const T*)x.data_ptr(), (const TW*)wc.data_ptr(),
                             (const TW*)bc.data_ptr(), (T*)y.data_ptr(), mean.data_ptr<float>(),
                             rstd.data_ptr<float>(), N, D, (float)eps, ys);
      } else if (silu)
        hipLaunchKernelGGL((ln_act_fwd_kernel<T, TW, true>), dim3((int)N), dim3(kBlock), 0, stream.stream(),
                           (const T*)x.data_ptr(), (const TW*)wc.data_ptr(), (const TW*)bc.data_ptr(),
                           (T*)y.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(), D, (float)eps, ys,
                           N <= 64);
      else
        hipLaunchKernelGGL((ln_act_fwd_kernel<T, TW, false>), dim3((int)N), dim3(kBlock), 0, stream.stream(),
                           (const T*)x.data_ptr(), (const TW*)wc.data_ptr(), (const TW*)bc.data_ptr(),
                           (T*)y.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(), D, (float)eps, ys,
                           N <= 64);
    });
  });
}

std::vector<torch::Tensor> ln_act_fwd(const torch::Tensor& x, const torch::Tensor& w, const torch::Tensor& b,
                                      double eps, bool silu) {
  CHECK_IN(x);
  TORCH_CHECK(x.dim() == 2, "ln_act_fwd expects [N, D]");
  long N = x.size(0);
  int D = (int)x.size(1);
  auto y = torch::empty_like(x);
  auto mean = torch::empty({N}, x.options().dtype(at::kFloat));
  auto rstd = torch::empty({N}, x.options().dtype(at::kFloat));
  ln_act_fwd_core(x, w, b, eps, silu, y, mean, rstd, (long)D);
  return {y, mean, rstd};
}

// Out-variant for the fused scan: y may be a row-strided 2-D view (a slice of
// a stacked [T, B, *] buffer); mean/rstd are caller-provided fp32 [N] slices.
torch::Tensor ln_act_fwd_o(const torch::Tensor& x, const torch::Tensor& w, const torch::Tensor& b, double eps,
                           bool silu, torch::Tensor y, torch::Tensor mean, torch::Tensor rstd) {
  CHECK_IN(x);
  TORCH_CHECK(x.dim() == 2 && y.dim() == 2 && y.stride(1) == 1, "ln_act_fwd_o shapes");
  TORCH_CHECK(y.scalar_type() == x.scalar_type() && mean.is_contiguous() && rstd.is_contiguous());
  ln_act_fwd_core(x, w, b, eps, silu, y, mean, rstd, y.stride(0));
  return y;
}

// Two-stage weight-grad reduction: a bounded grid of blocks strides the rows,
// each accumulating gw/gb in an LDS fp32 image of the row, with ONE atomicAdd
// per column per block at the end (instead of one per column per ROW, which
// serialized on atomics contention — measured 334 us/call on the DV3 bench).
template <typename T, typename TW, bool SILU>
__global__ void ln_act_bwd_kernel(const T* __restrict__ gy, const T* __restrict__ x, const TW* __restrict__ w,
                                  const TW* __restrict__ b, const float* __restrict__ mean,
                                  const float* __restrict__ rstd, T* __restrict__ gx, float* __restrict__ gw,
                                  float* __restrict__ gb, long N, int D, long gys, bool cached) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  float* lds = smem;          // 9 floats for block_sum
  float* gw_acc = smem + 32;  // [D] (block_sum2 scratch precedes)
  float* gb_acc = gw_acc + D; // [D]
  for (int j = threadIdx.x; j < D; j += blockDim.x) {
    gw_acc[j] = 0.f;
    gb_acc[j] = 0.f;
  }
  __syncthreads();
  // register-cached single-read path: xhat and post-SILU gz persist across
  // the reduction (no second read of x/gy, no SILU recompute); S1/S2 reduce
  // in one fused barrier round.  8 floats each covers D <= 2048 at 256 thr.
  // Scratch-backed caches (see ln_act_fwd_kernel): host gates `cached` by N
  // (measured: the behaviour MLP bwd at 16384 rows ran 93.9 us cached).
  if (cached && D <= 8 * (int)blockDim.x) {
    float cxh[8], cgz[8];
    for (long row = blockIdx.x; row < N; row += gridDim.x) {
      const T* xr = x + row * (long)D;
      const T* gr = gy + row * gys;
      T* gxr = gx + row * (long)D;
      const float m = mean[row], r = rstd[row];
      float s1 = 0.f, s2 = 0.f;
      int cnt = 0;
      for (int j = threadIdx.x; j < D; j += blockDim.x) {
        float xhat = (ld(xr, j) - m) * r;
        float gz = ld(gr, j);
        if (SILU) {
          float z = xhat * ld(w, j) + ld(b, j);
          float sig = 1.f / (1.f + texp<T>(-z));
          gz *= sig * (1.f + z * (1.f - sig));
        }
        cxh[cnt] = xhat;
        cgz[cnt] = gz;
        ++cnt;
        float gxhat = gz * ld(w, j);
        s1 += gxhat;
        s2 += gxhat * xhat;
      }
      block_sum2(s1, s2, lds);
      const float S1 = s1 / D, S2 = s2 / D;
      cnt = 0;
      for (int j = threadIdx.x; j < D; j += blockDim.x) {
        float xhat = cxh[cnt], gz = cgz[cnt];
        ++cnt;
        gw_acc[j] += gz * xhat;
        gb_acc[j] += gz;
        float gxhat = gz * ld(w, j);
        st(gxr, j, (gxhat - S1 - xhat * S2) * r);
      }
      __syncthreads();
    }
    for (int j = threadIdx.x; j < D; j += blockDim.x) {
      atomicAdd(&gw[j], gw_acc[j]);
      atomicAdd(&gb[j], gb_acc[j]);
    }
    return;
  }
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const T* xr = x + row * (long)D;
    const T* gr = gy + row * gys;
    T* gxr = gx + row * (long)D;
    const float m = mean[row], r = rstd[row];
    float s1 = 0.f, s2 = 0.f;
    for (int j = threadIdx.x; j < D; j += blockDim.x) {
      float xhat = (ld(xr, j) - m) * r;
      float gz = ld(gr, j);
      if (SILU) {
        float z = xhat * ld(w, j) + ld(b, j);
        float sig = 1.f / (1.f + texp<T>(-z));
        gz *= sig * (1.f + z * (1.f - sig));
      }
      float gxhat = gz * ld(w, j);
      s1 += gxhat;
      s2 += gxhat * xhat;
    }
    float S1 = block_sum(s1, lds) / D;
    __syncthreads();
    float S2 = block_sum(s2, lds) / D;
    for (int j = threadIdx.x; j < D; j += blockDim.x) {
      float xhat = (ld(xr, j) - m) * r;
      float gz = ld(gr, j);
      if (SILU) {
        float z = xhat * ld(w, j) + ld(b, j);
        float sig = 1.f / (1.f + texp<T>(-z));
        gz *= sig * (1.f + z * (1.f - sig));
      }
      gw_acc[j] += gz * xhat;
      gb_acc[j] += gz;
      float gxhat = gz * ld(w, j);
      st(gxr, j, (gxhat - S1 - xhat * S2) * r);
    }
    __syncthreads();
  }
  for (int j = threadIdx.x; j < D; j += blockDim.x) {
    atomicAdd(&gw[j], gw_acc[j]);
    atomicAdd(&gb[j], gb_acc[j]);
  }
}

// Wave-per-row backward for short rows; gw/gb accumulate in per-wave LDS
// slices, reduced block-wide and flushed with one atomicAdd per column per
// block.
template <typename T, typename TW, bool SILU>
__global__ void ln_act_bwd_small_kernel(const T* __restrict__ gy, const T* __restrict__ x,
                                        const TW* __restrict__ w, const TW* __restrict__ b,
                                        const float* __restrict__ mean, const float* __restrict__ rstd,
                                        T* __restrict__ gx, float* __restrict__ gw, float* __restrict__ gb,
                                        long N, int D, long gys) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  const int nw = blockDim.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  float* gw_acc = smem + (size_t)wid * 2 * D;      // per-wave [D]
  float* gb_acc = gw_acc + D;
  for (int j = lane; j < D; j += 64) {
    gw_acc[j] = 0.f;
    gb_acc[j] = 0.f;
  }
  const long wave = (long)blockIdx.x * nw + wid;
  const long nwaves = (long)gridDim.x * nw;
  for (long row = wave; row < N; row += nwaves) {
    const T* xr = x + row * (long)D;
    const T* gr = gy + row * gys;
    T* gxr = gx + row * (long)D;
    const float m = mean[row], r = rstd[row];
    float s1 = 0.f, s2 = 0.f;
    for (int j = lane; j < D; j += 64) {
      float xhat = (ld(xr, j) - m) * r;
      float gz = ld(gr, j);
      if (SILU) {
        float z = xhat * ld(w, j) + ld(b, j);
        float sig = 1.f / (1.f + texp<T>(-z));
        gz *= sig * (1.f + z * (1.f - sig));
      }
      float gxhat = gz * ld(w, j);
      s1 += gxhat;
      s2 += gxhat * xhat;
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      s1 += __shfl_xor(s1, off, 64);
      s2 += __shfl_xor(s2, off, 64);
    }
    float S1 = s1 / D, S2 = s2 / D;
    for (int j = lane; j < D; j += 64) {
      float xhat = (ld(xr, j) - m) * r;
      float gz = ld(gr, j);
      if (SILU) {
        float z = xhat * ld(w, j) + ld(b, j);
        float sig = 1.f / (1.f + texp<T>(-z));
        gz *= sig * (1.f + z * (1.f - sig));
      }
      gw_acc[j] += gz * xhat;
      gb_acc[j] += gz;
      float gxhat = gz * ld(w, j);
      st(gxr, j, (gxhat - S1 - xhat * S2) * r);
    }
  }
  __syncthreads();
  for (int j = threadIdx.x; j < D; j += blockDim.x) {
    float sw = 0.f, sb = 0.f;
    for (int k = 0; k < nw; ++k) {
      sw += smem[(size_t)k * 2 * D + j];
      sb += smem[(size_t)k * 2 * D + D + j];
    }
    atomicAdd(&gw[j], sw);
    atomicAdd(&gb[j], sb);
  }
}

void ln_act_bwd_core(const torch::Tensor& gy, const torch::Tensor& x, const torch::Tensor& w,
                     const torch::Tensor& b, const torch::Tensor& mean, const torch::Tensor& rstd, bool silu,
                     torch::Tensor& gx, torch::Tensor& gw, torch::Tensor& gb, long gys) {
  CHECK_IN(x);
  long N = x.size(0);
  int D = (int)x.size(1);
  auto wc = w.contiguous();
  auto bc = b.contiguous();
  {
    auto streamv = at::cuda::getCurrentCUDAStream();
    bool done = false;
    AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "ln_act_bwd_cl", [&] {
      using T = scalar_t;
      AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, wc.scalar_type(), "ln_act_bwd_cl_w", [&] {
        using TW = scalar_t;
        const int L = ln_cl_lanes<T>(D, gys);
        if (L) {
          if (silu)
            launch_ln_bwd_cl<T, TW, true>(L, N, streamv.stream(), (const T*)gy.data_ptr(),
                                          (const T*)x.data_ptr(), (const TW*)wc.data_ptr(),
                                          (const TW*)bc.data_ptr(), mean.data_ptr<float>(),
                                          rstd.data_ptr<float>(), (T*)gx.data_ptr(), gw.data_ptr<float>(),
                                          gb.data_ptr<float>(), D);
          else
            launch_ln_bwd_cl<T, TW, false>(L, N, streamv.stream(), (const T*)gy.data_ptr(),
                                           (const T*)x.data_ptr(), (const TW*)wc.data_ptr(),
                                           (const TW*)bc.data_ptr(), mean.data_ptr<float>(),
                                           rstd.data_ptr<float>(), (T*)gx.data_ptr(), gw.data_ptr<float>(),
                                           gb.data_ptr<float>(), D);
          done = true;
        }
      });
    });
    if (done) return;
  }
  if (D <= 256) {
    size_t shmem = (size_t)(kBlock >> 6) * 2 * D * sizeof(float);
    int blocks = (int)std::min((N + 3) / 4, (long)2048);
    auto stream2 = at::cuda::getCurrentCUDAStream();
    AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "ln_act_bwd_s", [&] {
      using T = scalar_t;
      AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, wc.scalar_type(), "ln_act_bwd_s_w", [&] {
        using TW = scalar_t;
        if (silu)
          hipLaunchKernelGGL((ln_act_bwd_small_kernel<T, TW, true>), dim3(blocks), dim3(kBlock), shmem,
                             stream2.stream(), (const T*)gy.data_ptr(), (const T*)x.data_ptr(),
                             (const TW*)wc.data_ptr(), (const TW*)bc.data_ptr(), mean.data_ptr<float>(),
                             rstd.data_ptr<float>(), (T*)gx.data_ptr(), gw.data_ptr<float>(),
                             gb.data_ptr<float>(), N, D, gys);
        else
          hipLaunchKernelGGL((ln_act_bwd_small_kernel<T, TW, false>), dim3(blocks), dim3(kBlock), shmem,
                             stream2.stream(), (const T*)gy.data_ptr(), (const T*)x.data_ptr(),
                             (const TW*)wc.data_ptr(), (const TW*)bc.data_ptr(), mean.data_ptr<float>(),
                             rstd.data_ptr<float>(), (T*)gx.data_ptr(), gw.data_ptr<float>(),
                             gb.data_ptr<float>(), N, D, gys);
      });
    });
    return;
  }
  size_t shmem = (32 + 2 * (size_t)D) * sizeof(float);
  TORCH_CHECK(shmem <= 160 * 1024, "ln_act_bwd: D too large for LDS accumulation");
  int blocks = (int)std::min(N, (long)512);
  auto stream = at::cuda::getCurrentCUDAStream();
  {
    bool done = false;
    AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "ln_act_bwd_v", [&] {
      using T = scalar_t;
      AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, wc.scalar_type(), "ln_act_bwd_v_w", [&] {
        using TW = scalar_t;
        constexpr int V = 16 / sizeof(T);
        int K = 0;
        if (N > 64 && gys % V == 0) {
          if (D % (2 * V) == 0 && D / (2 * V) <= 64)
            K = 2;
          else if (D % (4 * V) == 0 && D / (4 * V) <= 64)
            K = 4;
        }
        if (K) {
          // vectorized wave-per-row; bounded blocks keep the per-block
          // gw/gb atomic flush (blocks x D adds) off the critical path
          int vblocks = (int)std::min((N + 3) / 4, (long)768);
          if (silu) {
            if (K == 2)
              hipLaunchKernelGGL((ln_act_bwd_v_kernel<T, TW, true, 2>), dim3(vblocks), dim3(kBlock),
                                 shmem, stream.stream(), (const T*)gy.data_ptr(), (const T*)x.data_ptr(),
                                 (const TW*)wc.data_ptr(), (const TW*)bc.data_ptr(),
                                 mean.data_ptr<float>(), rstd.data_ptr<float>(), (T*)gx.data_ptr(),
                                 gw.data_ptr<float>(), gb.data_ptr<float>(), N, D, gys);
            else
              hipLaunchKernelGGL((ln_act_bwd_v_kernel<T, TW, true, 4>), dim3(vblocks), dim3(kBlock),
                                 shmem, stream.stream(), (const T*)gy.data_ptr(), (const T*)x.data_ptr(),
                                 (const TW*)wc.data_ptr(), (const TW*)bc.data_ptr(),
                                 mean.data_ptr<float>(), rstd.data_ptr<float>(), (T*)gx.data_ptr(),
                                 gw.data_ptr<float>(), gb.data_ptr<float>(), N, D, gys);
          } else {
            if (K == 2)
              hipLaunchKernelGGL((ln_act_bwd_v_kernel<T, TW, false, 2>), dim3(vblocks), dim3(kBlock),
                                 shmem, stream.stream(), (const T*)gy.data_ptr(), (const T*)x.data_ptr(),
                                 (const TW*)wc.data_ptr(), (const TW*)bc.data_ptr(),
                                 mean.data_ptr<float>(), rstd.data_ptr<float>(), (T*)gx.data_ptr(),
                                 gw.data_ptr<float>(), gb.data_ptr<float>(), N, D, gys);
            else
              hipLaunchKernelGGL((ln_act_bwd_v_kernel<T, TW, false, 4>), dim3(vblocks), dim3(kBlock),
                                 shmem, stream.stream(), (const T*)gy.data_ptr(), (const T*)x.data_ptr(),
                                 (const TW*)wc.data_ptr(), (const TW*)bc.data_ptr(),
                                 mean.data_ptr<float>(), rstd.data_ptr<float>(), (T*)gx.data_ptr(),
                                 gw.data_ptr<float>(), gb.data_ptr<float>(), N, D, gys);
          }
          done = true;
        }
      });
    });
    if (done) return;
  }
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "ln_act_bwd", [&] {
    using T = scalar_t;
    AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, wc.scalar_type(), "ln_act_bwd_w", [&] {
      using TW = scalar_t;
      if (silu)
        hipLaunchKernelGGL((ln_act_bwd_kernel<T, TW, true>), dim3(blocks), dim3(kBlock), shmem, stream.stream(),
                           (const T*)gy.data_ptr(), (const T*)x.data_ptr(), (const TW*)wc.data_ptr(),
                           (const TW*)bc.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(),
                           (T*)gx.data_ptr(), gw.data_ptr<float>(), gb.data_ptr<float>(), N, D, gys, N <= 64);
      else
        hipLaunchKernelGGL((ln_act_bwd_kernel<T, TW, false>), dim3(blocks), dim3(kBlock), shmem, stream.stream(),
                           (const T*)gy.data_ptr(), (const T*)x.data_ptr(), (const TW*)wc.data_ptr(),
                           (const TW*)bc.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(),
                           (T*)gx.data_ptr(), gw.data_ptr<float>(), gb.data_ptr<float>(), N, D, gys, N <= 64);
    });
  });
}

std::vector<torch::Tensor> ln_act_bwd(const torch::Tensor& gy, const torch::Tensor& x, const torch::Tensor& w,
                                      const torch::Tensor& b, const torch::Tensor& mean, const torch::Tensor& rstd,
                                      bool silu) {
  CHECK_IN(gy);
  CHECK_IN(x);
  int D = (int)x.size(1);
  auto gx = torch::empty_like(x);
  auto gw = torch::zeros({D}, x.options().dtype(at::kFloat));
  auto gb = torch::zeros({D}, x.options().dtype(at::kFloat));
  ln_act_bwd_core(gy, x, w, b, mean, rstd, silu, gx, gw, gb, (long)D);
  return {gx, gw.to(w.scalar_type()), gb.to(b.scalar_type())};
}

// Accumulate-variant for the fused scan backward: gy may be row-strided;
// gx is a caller-provided contiguous output; gw/gb are fp32 accumulators
// (zeroed once per scan by the caller) — no per-step zero-fill or cast.
torch::Tensor ln_act_bwd_acc(const torch::Tensor& gy, const torch::Tensor& x, const torch::Tensor& w,
                             const torch::Tensor& b, const torch::Tensor& mean, const torch::Tensor& rstd,
                             bool silu, torch::Tensor gx, torch::Tensor gw, torch::Tensor gb) {
  TORCH_CHECK(gy.dim() == 2 && gy.stride(1) == 1 && gx.is_contiguous(), "ln_act_bwd_acc shapes");
  TORCH_CHECK(gw.scalar_type() == at::kFloat && gb.scalar_type() == at::kFloat, "acc buffers must be fp32");
  ln_act_bwd_core(gy, x, w, b, mean, rstd, silu, gx, gw, gb, gy.stride(0));
  return gx;
}

// ---------------------------------------------------------------------------
// GRU gates: z = LN(y)*w+b; r,c,u = chunk(z,3); h' = σ(u-1)*tanh(σ(r)*c) + (1-σ(u-1))*h
// ---------------------------------------------------------------------------

template <typename T, typename TW>
__global__ void gru_gates_fwd_kernel(const T* __restrict__ y, const T* __restrict__ h, const TW* __restrict__ w,
                                     const TW* __restrict__ b, T* __restrict__ hout, float* __restrict__ mean_out,
                                     float* __restrict__ rstd_out, int H, float eps, long hs,
                                     T* __restrict__ hout2, long h2s, T* __restrict__ hout3, long h3s,
                                     bool cached) {
  __shared__ float lds[18];
  const long row = blockIdx.x;
  const int D = 3 * H;
  const T* yr = y + row * (long)D;
  const T* hr = h + row * hs;
  T* outr = hout + row * (long)H;
  T* outr2 = hout2 ? hout2 + row * h2s : nullptr;
  T* outr3 = hout3 ? hout3 + row * h3s : nullptr;
  // single-read register-cached path (latency-bound at scan batch sizes):
  // cache element j at index (j - threadIdx.x)/blockDim, so the gates pass
  // finds positions j / H+j / 2H+j when H is a blockDim multiple
  if (cached && D <= 16 * (int)blockDim.x && (H % (int)blockDim.x) == 0) {
    float cache[16];
    float s = 0.f, s2 = 0.f;
    int cnt = 0;
    for (int j = threadIdx.x; j < D; j += blockDim.x) {
      float v = ld(yr, j);
      cache[cnt++] = v;
      s += v;
      s2 += v * v;
    }
    block_sum2(s, s2, lds);
    float mean = s / D;
    float var = s2 / D - mean * mean;
    float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
    if (threadIdx.x == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    const int hstep = H / (int)blockDim.x;  // cache entries per H-section
    for (int i = 0; i < hstep; ++i) {
      int j = threadIdx.x + i * blockDim.x;
      float zr = ((cache[i] - mean) * rstd) * ld(w, j) + ld(b, j);
      float zc = ((cache[hstep + i] - mean) * rstd) * ld(w, H + j) + ld(b, H + j);
      float zu = ((cache[2 * hstep + i] - mean) * rstd) * ld(w, 2 * H + j) + ld(b, 2 * H + j);
      float r = 1.f / (1.f + texp<T>(-zr));
      float c = ttanh<T>(r * zc);
      float u = 1.f / (1.f + texp<T>(-(zu - 1.f)));
      float hv = u * c + (1.f - u) * ld(hr, j);
      st(outr, j, hv);
      if (outr2) st(outr2, j, hv);
      if (outr3) st(outr3, j, hv);
    }
    return;
  }
  float s = 0.f;
  for (int j = threadIdx.x; j < D; j += blockDim.x) s += ld(yr, j);
  float mean = block_sum(s, lds) / D;
  __syncthreads();
  float s2 = 0.f;
  for (int j = threadIdx.x; j < D; j += blockDim.x) {
    float d = ld(yr, j) - mean;
    s2 += d * d;
  }
  float var = block_sum(s2, lds) / D;
  float rstd = rsqrtf(var + eps);
  if (threadIdx.x == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  for (int j = threadIdx.x; j < H; j += blockDim.x) {
    float zr = ((ld(yr, j) - mean) * rstd) * ld(w, j) + ld(b, j);
    float zc = ((ld(yr, H + j) - mean) * rstd) * ld(w, H + j) + ld(b, H + j);
    float zu = ((ld(yr, 2 * H + j) - mean) * rstd) * ld(w, 2 * H + j) + ld(b, 2 * H + j);
    float r = 1.f / (1.f + texp<T>(-zr));
    float c = ttanh<T>(r * zc);
    float u = 1.f / (1.f + texp<T>(-(zu - 1.f)));
    float hv = u * c + (1.f - u) * ld(hr, j);
    st(outr, j, hv);
    if (outr2) st(outr2, j, hv);
    if (outr3) st(outr3, j, hv);
  }
}

// Column-chunked ("wide") variant for few rows x large H (the XL scan:
// B=16, H=4096 put only 16 workgroups on 256 CUs — measured 30-60 us/call
// while doing ~1 MB of traffic).  Grid (N, C): each block redundantly
// streams the full row for the LN statistics (reads are L2/MALL hits across
// the row's blocks) and then computes/writes only its H-chunk — same launch
// count, ~C x the parallelism.
template <typename T, typename TW>
__global__ void gru_gates_fwd_wide_kernel(const T* __restrict__ y, const T* __restrict__ h,
                                          const TW* __restrict__ w, const TW* __restrict__ b,
                                          T* __restrict__ hout, float* __restrict__ mean_out,
                                          float* __restrict__ rstd_out, int H, float eps, long hs,
                                          T* __restrict__ hout2, long h2s, T* __restrict__ hout3, long h3s) {
  __shared__ float lds[18];
  const long row = blockIdx.x;
  const int D = 3 * H;
  const T* yr = y + row * (long)D;
  const T* hr = h + row * hs;
  T* outr = hout + row * (long)H;
  T* outr2 = hout2 ? hout2 + row * h2s : nullptr;
  T* outr3 = hout3 ? hout3 + row * h3s : nullptr;
  float s = 0.f, s2 = 0.f;
  for (int j = threadIdx.x; j < D; j += blockDim.x) {
    float v = ld(yr, j);
    s += v;
    s2 += v * v;
  }
  block_sum2(s, s2, lds);
  const float mean = s / D;
  const float var = s2 / D - mean * mean;
  const float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
  if (blockIdx.y == 0 && threadIdx.x == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  const int W = (H + (int)gridDim.y - 1) / (int)gridDim.y;
  const int j0 = (int)blockIdx.y * W, j1 = min(j0 + W, H);
  for (int j = j0 + (int)threadIdx.x; j < j1; j += blockDim.x) {
    float zr = ((ld(yr, j) - mean) * rstd) * ld(w, j) + ld(b, j);
    float zc = ((ld(yr, H + j) - mean) * rstd) * ld(w, H + j) + ld(b, H + j);
    float zu = ((ld(yr, 2 * H + j) - mean) * rstd) * ld(w, 2 * H + j) + ld(b, 2 * H + j);
    float r = 1.f / (1.f + texp<T>(-zr));
    float c = ttanh<T>(r * zc);
    float u = 1.f / (1.f + texp<T>(-(zu - 1.f)));
    float hv = u * c + (1.f - u) * ld(hr, j);
    st(outr, j, hv);
    if (outr2) st(outr2, j, hv);
    if (outr3) st(outr3, j, hv);
  }
}

// Vectorized block-per-row forward for MANY rows (the imagination rollout:
// B*T=1024-16384 rows).  The generic kernel makes 3 scalar passes over the
// 3H row (42-55 us at [1024, 12288]); this one makes a single vectorized
// stats pass (E[x^2] form) plus one vectorized gates pass.  y/hout are
// contiguous and vector-loaded; h may be an unaligned strided slice of the
// stacked GRU-input buffer and stays scalar (1/4 of the read bytes).
template <typename T, typename TW>
__global__ void __launch_bounds__(kBlock) gru_gates_fwd_vec_kernel(
    const T* __restrict__ y, const T* __restrict__ h, const TW* __restrict__ w, const TW* __restrict__ b,
    T* __restrict__ hout, float* __restrict__ mean_out, float* __restrict__ rstd_out, long N, int H,
    float eps, long hs, T* __restrict__ hout2, long h2s, T* __restrict__ hout3, long h3s) {
  constexpr int V = 16 / sizeof(T);
  __shared__ float lds[18];
  const int D = 3 * H;
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const T* yr = y + row * (long)D;
    const T* hr = h + row * hs;
    T* outr = hout + row * (long)H;
    T* outr2 = hout2 ? hout2 + row * h2s : nullptr;
    T* outr3 = hout3 ? hout3 + row * h3s : nullptr;
    float s = 0.f, s2 = 0.f;
    for (int j0 = (int)threadIdx.x * V; j0 < D; j0 += (int)blockDim.x * V) {
      LnVec<T, V> xv;
      xv.u = *reinterpret_cast<const uint4*>(yr + j0);
#pragma unroll
      for (int e = 0; e < V; ++e) {
        float v = ld(xv.e, e);
        s += v;
        s2 += v * v;
      }
    }
    block_sum2(s, s2, lds);
    const float mean = s / D;
    const float var = s2 / D - mean * mean;
    const float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
    if (threadIdx.x == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    for (int j0 = (int)threadIdx.x * V; j0 < H; j0 += (int)blockDim.x * V) {
      LnVec<T, V> yr_, yc_, yu_, ov;
      yr_.u = *reinterpret_cast<const uint4*>(yr + j0);
      yc_.u = *reinterpret_cast<const uint4*>(yr + H + j0);
      yu_.u = *reinterpret_cast<const uint4*>(yr + 2 * H + j0);
#pragma unroll
      for (int e = 0; e < V; ++e) {
        const int j = j0 + e;
        float zr = ((ld(yr_.e, e) - mean) * rstd) * ld(w, j) + ld(b, j);
        float zc = ((ld(yc_.e, e) - mean) * rstd) * ld(w, H + j) + ld(b, H + j);
        float zu = ((ld(yu_.e, e) - mean) * rstd) * ld(w, 2 * H + j) + ld(b, 2 * H + j);
        float r = 1.f / (1.f + texp<T>(-zr));
        float c = ttanh<T>(r * zc);
        float u = 1.f / (1.f + texp<T>(-(zu - 1.f)));
        float hv = u * c + (1.f - u) * ld(hr, j);
        st(ov.e, e, hv);
        if (outr2) st(outr2, j, hv);
        if (outr3) st(outr3, j, hv);
      }
      *reinterpret_cast<uint4*>(outr + j0) = ov.u;
    }
    __syncthreads();
  }
}

void gru_gates_fwd_core(const torch::Tensor& y, const torch::Tensor& h, const torch::Tensor& w,
                        const torch::Tensor& b, double eps, torch::Tensor& hout, torch::Tensor& mean,
                        torch::Tensor& rstd, long hs, void* hout2, long h2s, void* hout3 = nullptr,
                        long h3s = 0) {
  CHECK_IN(y);
  TORCH_CHECK(y.dim() == 2 && h.dim() == 2 && y.size(1) == 3 * h.size(1), "gru_gates_fwd shapes");
  long N = y.size(0);
  int H = (int)h.size(1);
  auto wc = w.contiguous();
  auto bc = b.contiguous();
  auto stream = at::cuda::getCurrentCUDAStream();
  // wide dispatch: few rows x large H leave the chip idle on the row-per-
  // block kernel; chunk columns so N*C workgroups >= ~384
  const bool wide = N <= 64 && H >= 2048;
  const int C = wide ? std::min<int>(std::max<int>(1, 512 / (int)N), (H + 127) / 128) : 1;
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, y.scalar_type(), "gru_gates_fwd", [&] {
    using T = scalar_t;
    AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, wc.scalar_type(), "gru_gates_fwd_w", [&] {
      using TW = scalar_t;
      if (wide)
        hipLaunchKernelGGL((gru_gates_fwd_wide_kernel<T, TW>), dim3((int)N, C), dim3(kBlock), 0,
                           stream.stream(), (const T*)y.data_ptr(), (const T*)h.data_ptr(),
                           (const TW*)wc.data_ptr(), (const TW*)bc.data_ptr(), (T*)hout.data_ptr(),
                           mean.data_ptr<float>(), rstd.data_ptr<float>(), H, (float)eps, hs, (T*)hout2,
                           h2s, (T*)hout3, h3s);
      else if (N > 64 && H % (16 / (int)sizeof(T)) == 0)
        hipLaunchKernelGGL((gru_gates_fwd_vec_kernel<T, TW>), dim3((int)std::min(N, (long)2048)),
                           dim3(kBlock), 0, stream.stream(), (const T*)y.data_ptr(),
                           (const T*)h.data_ptr(), (const TW*)wc.data_ptr(), (const TW*)bc.data_ptr(),
                           (T*)hout.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(), N, H,
                           (float)eps, hs, (T*)hout2, h2s, (T*)hout3, h3s);
      else
        hipLaunchKernelGGL((gru_gates_fwd_kernel<T, TW>), dim3((int)N), dim3(kBlock), 0, stream.stream(),
                           (const T*)y.data_ptr(), (const T*)h.data_ptr(), (const TW*)wc.data_ptr(),
                           (const TW*)bc.data_ptr(), (T*)hout.data_ptr(), mean.data_ptr<float>(),
                           rstd.data_ptr<float>(), H, (float)eps, hs, (T*)hout2, h2s, (T*)hout3, h3s,
                           N <= 64);
    });
  });
}

std::vector<torch::Tensor> gru_gates_fwd(const torch::Tensor& y, const torch::Tensor& h, const torch::Tensor& w,
                                         const torch::Tensor& b, double eps) {
  CHECK_IN(h);
  auto hout = torch::empty_like(h);
  auto mean = torch::empty({y.size(0)}, y.options().dtype(at::kFloat));
  auto rstd = torch::empty({y.size(0)}, y.options().dtype(at::kFloat));
  gru_gates_fwd_core(y, h, w, b, eps, hout, mean, rstd, h.size(1), nullptr, 0);
  return {hout, mean, rstd};
}

// Out-variant for the fused scan: hprev may be row-strided (a [:, :H] slice of
// the stacked GRU-input buffer); h is written both to the contiguous hout
// (next-step input / output sequence) and to a strided hout2 slice of the
// stacked representation-input buffer.  mean/rstd are caller-provided slices.
void gru_gates_fwd_o(const torch::Tensor& y, const torch::Tensor& h, const torch::Tensor& w,
                     const torch::Tensor& b, double eps, torch::Tensor hout, torch::Tensor hout2,
                     torch::Tensor mean, torch::Tensor rstd,
                     const c10::optional<torch::Tensor>& hout3 = c10::nullopt) {
  TORCH_CHECK(h.dim() == 2 && h.stride(1) == 1 && hout.is_contiguous(), "gru_gates_fwd_o shapes");
  TORCH_CHECK(hout2.dim() == 2 && hout2.stride(1) == 1 && hout2.scalar_type() == hout.scalar_type());
  void* h3 = nullptr;
  long h3s = 0;
  if (hout3.has_value()) {
    TORCH_CHECK(hout3->dim() == 2 && hout3->stride(1) == 1 && hout3->scalar_type() == hout.scalar_type());
    h3 = hout3->data_ptr();
    h3s = hout3->stride(0);
  }
  gru_gates_fwd_core(y, h, w, b, eps, hout, mean, rstd, h.stride(0), hout2.data_ptr(), hout2.stride(0),
                     h3, h3s);
}

template <typename T, typename TW>
__global__ void gru_gates_bwd_kernel(const T* __restrict__ gh, const T* __restrict__ gh2,
                                     const T* __restrict__ gh3, long gh3s, const T* __restrict__ y,
                                     const T* __restrict__ h, const TW* __restrict__ w, const TW* __restrict__ b,
                                     const float* __restrict__ mean, const float* __restrict__ rstd,
                                     T* __restrict__ gy, T* __restrict__ ghprev, float* __restrict__ gw,
                                     float* __restrict__ gb, long N, int H, long hs, bool cached) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  float* lds = smem;
  float* gw_acc = smem + 32;       // [3H] (block_sum2 scratch precedes)
  float* gb_acc = gw_acc + 3 * H;  // [3H]
  const int D = 3 * H;
  for (int j = threadIdx.x; j < D; j += blockDim.x) {
    gw_acc[j] = 0.f;
    gb_acc[j] = 0.f;
  }
  __syncthreads();
  // register-cached single-read path: xhat and gz live in registers across
  // the LN-backward reduction, so the second pass re-reads nothing and gy is
  // written once with its final value (the generic path stores gz into gy
  // and re-reads it); S1/S2 reduce together in one barrier round
  const int hstep = H / (int)blockDim.x;
  if (cached && (H % (int)blockDim.x) == 0 && hstep <= 4) {
    float cxh[12], cgz[12];
    for (long row = blockIdx.x; row < N; row += gridDim.x) {
      const T* yr = y + row * (long)D;
      const T* hr = h + row * hs;
      const T* ghr = gh + row * (long)H;
      const T* gh2r = gh2 ? gh2 + row * (long)H : nullptr;
      const T* gh3r = gh3 ? gh3 + row * gh3s : nullptr;
      T* gyr = gy + row * (long)D;
      T* ghp = ghprev + row * (long)H;
      const float m = mean[row], rs = rstd[row];
      float s1 = 0.f, s2 = 0.f;
      for (int i = 0; i < hstep; ++i) {
        int j = threadIdx.x + i * blockDim.x;
        float xh_r = (ld(yr, j) - m) * rs;
        float xh_c = (ld(yr, H + j) - m) * rs;
        float xh_u = (ld(yr, 2 * H + j) - m) * rs;
        float zr = xh_r * ld(w, j) + ld(b, j);
        float zc = xh_c * ld(w, H + j) + ld(b, H + j);
        float zu = xh_u * ld(w, 2 * H + j) + ld(b, 2 * H + j);
        float r = 1.f / (1.f + texp<T>(-zr));
        float rc = r * zc;
        float c = ttanh<T>(rc);
        float u = 1.f / (1.f + texp<T>(-(zu - 1.f)));
        float g = ld(ghr, j) + (gh2r ? ld(gh2r, j) : 0.f) + (gh3r ? ld(gh3r, j) : 0.f);
        float gu = g * (c - ld(hr, j));
        float gc = g * u;
        float gzu = gu * u * (1.f - u);
        float grc = gc * (1.f - c * c);
        float gzc = grc * r;
        float gr = grc * zc;
        float gzr = gr * r * (1.f - r);
        st(ghp, j, g * (1.f - u));
        cxh[i] = xh_r;
        cxh[hstep + i] = xh_c;
        cxh[2 * hstep + i] = xh_u;
        cgz[i] = gzr;
        cgz[hstep + i] = gzc;
        cgz[2 * hstep + i] = gzu;
        gw_acc[j] += gzr * xh_r;
        gb_acc[j] += gzr;
        gw_acc[H + j] += gzc * xh_c;
        gb_acc[H + j] += gzc;
        gw_acc[2 * H + j] += gzu * xh_u;
        gb_acc[2 * H + j] += gzu;
        float gxh_r = gzr * ld(w, j);
        float gxh_c = gzc * ld(w, H + j);
        float gxh_u = gzu * ld(w, 2 * H + j);
        s1 += gxh_r + gxh_c + gxh_u;
        s2 += gxh_r * xh_r + gxh_c * xh_c + gxh_u * xh_u;
      }
      block_sum2(s1, s2, lds);
      const float S1 = s1 / D, S2 = s2 / D;
      for (int sec = 0; sec < 3; ++sec)
        for (int i = 0; i < hstep; ++i) {
          int j = sec * H + threadIdx.x + i * blockDim.x;
          float gxhat = cgz[sec * hstep + i] * ld(w, j);
          st(gyr, j, (gxhat - S1 - cxh[sec * hstep + i] * S2) * rs);
        }
      __syncthreads();
    }
    for (int j = threadIdx.x; j < D; j += blockDim.x) {
      atomicAdd(&gw[j], gw_acc[j]);
      atomicAdd(&gb[j], gb_acc[j]);
    }
    return;
  }
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const T* yr = y + row * (long)D;
    const T* hr = h + row * hs;
    const T* ghr = gh + row * (long)H;
    const T* gh2r = gh2 ? gh2 + row * (long)H : nullptr;
    const T* gh3r = gh3 ? gh3 + row * gh3s : nullptr;
    T* gyr = gy + row * (long)D;
    T* ghp = ghprev + row * (long)H;
    const float m = mean[row], rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
    for (int j = threadIdx.x; j < H; j += blockDim.x) {
      float xh_r = (ld(yr, j) - m) * rs;
      float xh_c = (ld(yr, H + j) - m) * rs;
      float xh_u = (ld(yr, 2 * H + j) - m) * rs;
      float zr = xh_r * ld(w, j) + ld(b, j);
      float zc = xh_c * ld(w, H + j) + ld(b, H + j);
      float zu = xh_u * ld(w, 2 * H + j) + ld(b, 2 * H + j);
      float r = 1.f / (1.f + texp<T>(-zr));
      float rc = r * zc;
      float c = ttanh<T>(rc);
      float u = 1.f / (1.f + texp<T>(-(zu - 1.f)));
      float g = ld(ghr, j) + (gh2r ? ld(gh2r, j) : 0.f) + (gh3r ? ld(gh3r, j) : 0.f);
      float gu = g * (c - ld(hr, j));
      float gc = g * u;
      float gzu = gu * u * (1.f - u);
      float grc = gc * (1.f - c * c);
      float gzc = grc * r;
      float gr = grc * zc;
      float gzr = gr * r * (1.f - r);
      st(ghp, j, g * (1.f - u));
      // store gz temporarily in gy (pre-LN-backward)
      st(gyr, j, gzr);
      st(gyr, H + j, gzc);
      st(gyr, 2 * H + j, gzu);
      gw_acc[j] += gzr * xh_r;
      gb_acc[j] += gzr;
      gw_acc[H + j] += gzc * xh_c;
      gb_acc[H + j] += gzc;
      gw_acc[2 * H + j] += gzu * xh_u;
      gb_acc[2 * H + j] += gzu;
      float gxh_r = gzr * ld(w, j);
      float gxh_c = gzc * ld(w, H + j);
      float gxh_u = gzu * ld(w, 2 * H + j);
      s1 += gxh_r + gxh_c + gxh_u;
      s2 += gxh_r * xh_r + gxh_c * xh_c + gxh_u * xh_u;
    }
    float S1 = block_sum(s1, lds) / D;
    __syncthreads();
    float S2 = block_sum(s2, lds) / D;
    __syncthreads();
    for (int j = threadIdx.x; j < D; j += blockDim.x) {
      float xhat = (ld(yr, j) - m) * rs;
      float gz = ld(gyr, j);
      float gxhat = gz * ld(w, j);
      st(gyr, j, (gxhat - S1 - xhat * S2) * rs);
    }
    __syncthreads();
  }
  for (int j = threadIdx.x; j < D; j += blockDim.x) {
    atomicAdd(&gw[j], gw_acc[j]);
    atomicAdd(&gb[j], gb_acc[j]);
  }
}

// Two-stage wide backward: the single wide kernel redundantly recomputed the
// full-row gate math in every one of its C chunk blocks to form S1/S2
// (measured 31.7 us at H=4096).  Stage 1 computes each chunk's partial
// s1/s2 once (plain stores to a [N, C, 2] workspace — no atomics), stage 2
// sums the C partials and does the chunk's write pass.  Total gate math is
// 2x the minimum instead of C x; two ~5 us launches replace one 32 us one.
template <typename T, typename TW>
__device__ __forceinline__ void gru_col_bwd(const T* yr, const T* hr, const T* ghr, const T* gh2r,
                                            const T* gh3r, const TW* w, const TW* b, float m, float rs,
                                            int H, int j, float& gzr, float& gzc, float& gzu,
                                            float& ghp_v, float& xh_r, float& xh_c, float& xh_u) {
  xh_r = (ld(yr, j) - m) * rs;
  xh_c = (ld(yr, H + j) - m) * rs;
  xh_u = (ld(yr, 2 * H + j) - m) * rs;
  float zr = xh_r * ld(w, j) + ld(b, j);
  float zc = xh_c * ld(w, H + j) + ld(b, H + j);
  float zu = xh_u * ld(w, 2 * H + j) + ld(b, 2 * H + j);
  float r = 1.f / (1.f + texp<T>(-zr));
  float c = ttanh<T>(r * zc);
  float u = 1.f / (1.f + texp<T>(-(zu - 1.f)));
  float g = ld(ghr, j) + (gh2r ? ld(gh2r, j) : 0.f) + (gh3r ? ld(gh3r, j) : 0.f);
  float gu = g * (c - ld(hr, j));
  float gc = g * u;
  gzu = gu * u * (1.f - u);
  float grc = gc * (1.f - c * c);
  gzc = grc * r;
  gzr = grc * zc * r * (1.f - r);
  ghp_v = g * (1.f - u);
}

template <typename T, typename TW>
__global__ void __launch_bounds__(kBlock) gru_gates_bwd_stats_kernel(
    const T* __restrict__ gh, const T* __restrict__ gh2, const T* __restrict__ gh3, long gh3s,
    const T* __restrict__ y, const T* __restrict__ h, const TW* __restrict__ w, const TW* __restrict__ b,
    const float* __restrict__ mean, const float* __restrict__ rstd, float* __restrict__ spart, int H,
    long hs) {
  __shared__ float lds[18];
  const long row = blockIdx.x;
  const int C = (int)gridDim.y;
  const int W = (H + C - 1) / C;
  const int j0 = (int)blockIdx.y * W, j1 = min(j0 + W, H);
  const T* yr = y + row * (long)(3 * H);
  const T* hr = h + row * hs;
  const T* ghr = gh + row * (long)H;
  const T* gh2r = gh2 ? gh2 + row * (long)H : nullptr;
  const T* gh3r = gh3 ? gh3 + row * gh3s : nullptr;
  const float m = mean[row], rs = rstd[row];
  float s1 = 0.f, s2 = 0.f;
  for (int j = j0 + (int)threadIdx.x; j < j1; j += blockDim.x) {
    float gzr, gzc, gzu, ghp_v, xh_r, xh_c, xh_u;
    gru_col_bwd(yr, hr, ghr, gh2r, gh3r, w, b, m, rs, H, j, gzr, gzc, gzu, ghp_v, xh_r, xh_c, xh_u);
    float gxh_r = gzr * ld(w, j);
    float gxh_c = gzc * ld(w, H + j);
    float gxh_u = gzu * ld(w, 2 * H + j);
    s1 += gxh_r + gxh_c + gxh_u;
    s2 += gxh_r * xh_r + gxh_c * xh_c + gxh_u * xh_u;
  }
  block_sum2(s1, s2, lds);
  if (threadIdx.x == 0) {
    spart[(row * C + blockIdx.y) * 2] = s1;
    spart[(row * C + blockIdx.y) * 2 + 1] = s2;
  }
}

template <typename T, typename TW>
__global__ void __launch_bounds__(kBlock) gru_gates_bwd_apply_kernel(
    const T* __restrict__ gh, const T* __restrict__ gh2, const T* __restrict__ gh3, long gh3s,
    const T* __restrict__ y, const T* __restrict__ h, const TW* __restrict__ w, const TW* __restrict__ b,
    const float* __restrict__ mean, const float* __restrict__ rstd, const float* __restrict__ spart,
    T* __restrict__ gy, T* __restrict__ ghprev, float* __restrict__ gw, float* __restrict__ gb, int H,
    long hs) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  const long row = blockIdx.x;
  const int C = (int)gridDim.y;
  const int W = (H + C - 1) / C;
  float* gw_acc = smem;           // [3W]
  float* gb_acc = gw_acc + 3 * W; // [3W]
  const int D = 3 * H;
  const int j0 = (int)blockIdx.y * W, j1 = min(j0 + W, H);
  for (int j = threadIdx.x; j < 3 * W; j += blockDim.x) {
    gw_acc[j] = 0.f;
    gb_acc[j] = 0.f;
  }
  __syncthreads();
  float S1 = 0.f, S2 = 0.f;
  for (int c = 0; c < C; ++c) {
    S1 += spart[(row * C + c) * 2];
    S2 += spart[(row * C + c) * 2 + 1];
  }
  S1 /= D;
  S2 /= D;
  const T* yr = y + row * (long)D;
  const T* hr = h + row * hs;
  const T* ghr = gh + row * (long)H;
  const T* gh2r = gh2 ? gh2 + row * (long)H : nullptr;
  const T* gh3r = gh3 ? gh3 + row * gh3s : nullptr;
  T* gyr = gy + row * (long)D;
  T* ghp = ghprev + row * (long)H;
  const float m = mean[row], rs = rstd[row];
  for (int j = j0 + (int)threadIdx.x; j < j1; j += blockDim.x) {
    float gzr, gzc, gzu, ghp_v, xh_r, xh_c, xh_u;
    gru_col_bwd(yr, hr, ghr, gh2r, gh3r, w, b, m, rs, H, j, gzr, gzc, gzu, ghp_v, xh_r, xh_c, xh_u);
    st(ghp, j, ghp_v);
    const int jc = j - j0;
    gw_acc[jc] += gzr * xh_r;
    gb_acc[jc] += gzr;
    gw_acc[W + jc] += gzc * xh_c;
    gb_acc[W + jc] += gzc;
    gw_acc[2 * W + jc] += gzu * xh_u;
    gb_acc[2 * W + jc] += gzu;
    st(gyr, j, (gzr * ld(w, j) - S1 - xh_r * S2) * rs);
    st(gyr, H + j, (gzc * ld(w, H + j) - S1 - xh_c * S2) * rs);
    st(gyr, 2 * H + j, (gzu * ld(w, 2 * H + j) - S1 - xh_u * S2) * rs);
  }
  __syncthreads();
  for (int jc = threadIdx.x; jc < j1 - j0; jc += blockDim.x) {
    atomicAdd(&gw[j0 + jc], gw_acc[jc]);
    atomicAdd(&gb[j0 + jc], gb_acc[jc]);
    atomicAdd(&gw[H + j0 + jc], gw_acc[W + jc]);
    atomicAdd(&gb[H + j0 + jc], gb_acc[W + jc]);
    atomicAdd(&gw[2 * H + j0 + jc], gw_acc[2 * W + jc]);
    atomicAdd(&gb[2 * H + j0 + jc], gb_acc[2 * W + jc]);
  }
}

void gru_gates_bwd_core(const torch::Tensor& gh, const void* gh2, const void* gh3, long gh3s,
                        const torch::Tensor& y, const torch::Tensor& h, const torch::Tensor& w,
                        const torch::Tensor& b, const torch::Tensor& mean, const torch::Tensor& rstd,
                        torch::Tensor& gy, torch::Tensor& ghprev, torch::Tensor& gw, torch::Tensor& gb,
                        long hs) {
  CHECK_IN(gh);
  CHECK_IN(y);
  long N = y.size(0);
  int H = (int)(y.size(1) / 3);
  auto wc = w.contiguous();
  auto bc = b.contiguous();
  auto stream = at::cuda::getCurrentCUDAStream();
  const bool wide = N <= 64 && H >= 2048;
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, y.scalar_type(), "gru_gates_bwd", [&] {
    using T = scalar_t;
    AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, wc.scalar_type(), "gru_gates_bwd_w", [&] {
      using TW = scalar_t;
      if (wide) {
        const int C = std::min<int>(std::max<int>(1, 512 / (int)N), (H + 127) / 128);
        const int W = (H + C - 1) / C;
        auto spart = torch::empty({N * (long)C * 2}, y.options().dtype(at::kFloat));
        hipLaunchKernelGGL((gru_gates_bwd_stats_kernel<T, TW>), dim3((int)N, C), dim3(kBlock), 0,
                           stream.stream(), (const T*)gh.data_ptr(), (const T*)gh2, (const T*)gh3, gh3s,
                           (const T*)y.data_ptr(), (const T*)h.data_ptr(), (const TW*)wc.data_ptr(),
                           (const TW*)bc.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(),
                           spart.data_ptr<float>(), H, hs);
        size_t shmem = 6 * (size_t)W * sizeof(float);
        hipLaunchKernelGGL((gru_gates_bwd_apply_kernel<T, TW>), dim3((int)N, C), dim3(kBlock), shmem,
                           stream.stream(), (const T*)gh.data_ptr(), (const T*)gh2, (const T*)gh3, gh3s,
                           (const T*)y.data_ptr(), (const T*)h.data_ptr(), (const TW*)wc.data_ptr(),
                           (const TW*)bc.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(),
                           spart.data_ptr<float>(), (T*)gy.data_ptr(), (T*)ghprev.data_ptr(),
                           gw.data_ptr<float>(), gb.data_ptr<float>(), H, hs);
        return;
      }
      size_t shmem = (32 + 6 * (size_t)H) * sizeof(float);
      TORCH_CHECK(shmem <= 160 * 1024, "gru_gates_bwd: H too large for LDS accumulation");
      int blocks = (int)std::min(N, (long)512);
      hipLaunchKernelGGL((gru_gates_bwd_kernel<T, TW>), dim3(blocks), dim3(kBlock), shmem, stream.stream(),
                         (const T*)gh.data_ptr(), (const T*)gh2, (const T*)gh3, gh3s,
                         (const T*)y.data_ptr(), (const T*)h.data_ptr(),
                         (const TW*)wc.data_ptr(), (const TW*)bc.data_ptr(), mean.data_ptr<float>(),
                         rstd.data_ptr<float>(), (T*)gy.data_ptr(), (T*)ghprev.data_ptr(), gw.data_ptr<float>(),
                         gb.data_ptr<float>(), N, H, hs, N <= 64);
    });
  });
}

std::vector<torch::Tensor> gru_gates_bwd(const torch::Tensor& gh, const torch::Tensor& y, const torch::Tensor& h,
                                         const torch::Tensor& w, const torch::Tensor& b, const torch::Tensor& mean,
                                         const torch::Tensor& rstd) {
  CHECK_IN(h);
  int H = (int)h.size(1);
  auto gy = torch::empty_like(y);
  auto ghprev = torch::empty_like(h);
  auto gw = torch::zeros({3 * H}, y.options().dtype(at::kFloat));
  auto gb = torch::zeros({3 * H}, y.options().dtype(at::kFloat));
  gru_gates_bwd_core(gh, nullptr, nullptr, 0, y, h, w, b, mean, rstd, gy, ghprev, gw, gb, (long)H);
  return {gy, ghprev, gw.to(w.scalar_type()), gb.to(b.scalar_type())};
}

// Accumulate-variant for the fused scan backward: hprev may be row-strided,
// gy/ghprev are caller-provided outputs, gw/gb fp32 accumulators.  gh2
// (contiguous) and gh3 (row-strided, e.g. a [:, :H] slice) are optional
// additional incoming-gradient terms summed with gh inside the kernel.
void gru_gates_bwd_acc(const torch::Tensor& gh, const c10::optional<torch::Tensor>& gh2,
                       const c10::optional<torch::Tensor>& gh3, const torch::Tensor& y, const torch::Tensor& h,
                       const torch::Tensor& w, const torch::Tensor& b, const torch::Tensor& mean,
                       const torch::Tensor& rstd, torch::Tensor gy, torch::Tensor ghprev, torch::Tensor gw,
                       torch::Tensor gb) {
  TORCH_CHECK(h.dim() == 2 && h.stride(1) == 1 && gy.is_contiguous() && ghprev.is_contiguous(),
              "gru_gates_bwd_acc shapes");
  TORCH_CHECK(gw.scalar_type() == at::kFloat && gb.scalar_type() == at::kFloat, "acc buffers must be fp32");
  const void* p2 = gh2.has_value() ? gh2->data_ptr() : nullptr;
  const void* p3 = gh3.has_value() ? gh3->data_ptr() : nullptr;
  long s3 = gh3.has_value() ? gh3->stride(0) : 0;
  gru_gates_bwd_core(gh, p2, p3, s3, y, h, w, b, mean, rstd, gy, ghprev, gw, gb, h.stride(0));
}

// ---------------------------------------------------------------------------
// reverse scans (fp32, [T, B] layout, lane-per-column)
// ---------------------------------------------------------------------------

__global__ void gae_scan_kernel(const float* __restrict__ rewards, const float* __restrict__ values,
                                const float* __restrict__ dones, const float* __restrict__ next_value,
                                float* __restrict__ adv, int T, long B, float gamma, float lam) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < B; i += (long)gridDim.x * blockDim.x) {
    float lastgaelam = 0.f;
    float nnt = 1.f - dones[(long)(T - 1) * B + i];
    float nv = next_value[i];
    for (int t = T - 1; t >= 0; --t) {
      if (t < T - 1) {
        nnt = 1.f - dones[(long)t * B + i];
        nv = values[(long)(t + 1) * B + i];
      }
      float delta = rewards[(long)t * B + i] + gamma * nv * nnt - values[(long)t * B + i];
      lastgaelam = delta + gamma * lam * nnt * lastgaelam;
      adv[(long)t * B + i] = lastgaelam;
    }
  }
}

torch::Tensor gae_scan(const torch::Tensor& rewards, const torch::Tensor& values, const torch::Tensor& dones,
                       const torch::Tensor& next_value, double gamma, double lam) {
  CHECK_IN(rewards);
  int T = (int)rewards.size(0);
  long B = rewards.numel() / T;
  auto adv = torch::empty_like(rewards);
  int blocks = (int)std::min((B + kBlock - 1) / kBlock, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(gae_scan_kernel, dim3(blocks), dim3(kBlock), 0, stream.stream(), rewards.data_ptr<float>(),
                     values.data_ptr<float>(), dones.data_ptr<float>(), next_value.data_ptr<float>(),
                     adv.data_ptr<float>(), T, B, (float)gamma, (float)lam);
  return adv;
}

__global__ void lambda_fwd_kernel(const float* __restrict__ r, const float* __restrict__ nv,
                                  const float* __restrict__ c, float* __restrict__ out, int T, long B, float lam) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < B; i += (long)gridDim.x * blockDim.x) {
    float nxt = nv[(long)(T - 1) * B + i];
    for (int t = T - 1; t >= 0; --t) {
      nxt = r[(long)t * B + i] + c[(long)t * B + i] * ((1.f - lam) * nv[(long)t * B + i] + lam * nxt);
      out[(long)t * B + i] = nxt;
    }
  }
}

torch::Tensor lambda_scan_fwd(const torch::Tensor& r, const torch::Tensor& nv, const torch::Tensor& c, double lam) {
  CHECK_IN(r);
  int T = (int)r.size(0);
  long B = r.numel() / T;
  auto out = torch::empty_like(r);
  int blocks = (int)std::min((B + kBlock - 1) / kBlock, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(lambda_fwd_kernel, dim3(blocks), dim3(kBlock), 0, stream.stream(), r.data_ptr<float>(),
                     nv.data_ptr<float>(), c.data_ptr<float>(), out.data_ptr<float>(), T, B, (float)lam);
  return out;
}

__global__ void lambda_bwd_kernel(const float* __restrict__ gy, const float* __restrict__ c, float* __restrict__ gr,
                                  float* __restrict__ gnv, int T, long B, float lam) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < B; i += (long)gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int t = 0; t < T; ++t) {
      float prev_c = (t > 0) ? c[(long)(t - 1) * B + i] : 0.f;
      acc = gy[(long)t * B + i] + ((t > 0) ? prev_c * lam * acc : 0.f);
      gr[(long)t * B + i] = acc;
      gnv[(long)t * B + i] = acc * c[(long)t * B + i] * (1.f - lam);
    }
    long last = (long)(T - 1) * B + i;
    gnv[last] += acc * c[last] * lam;
  }
}

std::vector<torch::Tensor> lambda_scan_bwd(const torch::Tensor& gy, const torch::Tensor& c, double lam) {
  CHECK_IN(gy);
  int T = (int)gy.size(0);
  long B = gy.numel() / T;
  auto gr = torch::empty_like(gy);
  auto gnv = torch::empty_like(gy);
  int blocks = (int)std::min((B + kBlock - 1) / kBlock, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(lambda_bwd_kernel, dim3(blocks), dim3(kBlock), 0, stream.stream(), gy.data_ptr<float>(),
                     c.data_ptr<float>(), gr.data_ptr<float>(), gnv.data_ptr<float>(), T, B, (float)lam);
  return {gr, gnv};
}

// ---------------------------------------------------------------------------
// multi-tensor Adam / EMA
// ---------------------------------------------------------------------------

template <typename T>
__global__ void adam_kernel(T* __restrict__ p, const T* __restrict__ g, float* __restrict__ m, float* __restrict__ v,
                            long n, float lr, float b1, float b2, float eps, float wd, float bc1, float bc2) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n; i += (long)gridDim.x * blockDim.x) {
    float gf = ld(g, i);
    float pf = ld(p, i);
    if (wd != 0.f) gf += wd * pf;
    float mi = b1 * m[i] + (1.f - b1) * gf;
    float vi = b2 * v[i] + (1.f - b2) * gf * gf;
    m[i] = mi;
    v[i] = vi;
    float upd = (mi / bc1) / (sqrtf(vi / bc2) + eps);
    st(p, i, pf - lr * upd);
  }
}

void adam_step(std::vector<torch::Tensor> params, std::vector<torch::Tensor> grads, std::vector<torch::Tensor> ms,
               std::vector<torch::Tensor> vs, double lr, double b1, double b2, double eps, double wd, double bc1,
               double bc2) {
  auto stream = at::cuda::getCurrentCUDAStream();
  for (size_t k = 0; k < params.size(); ++k) {
    auto& p = params[k];
    long n = p.numel();
    if (n == 0) continue;
    int blocks = (int)std::min((n + kBlock - 1) / kBlock, (long)2048);
    auto gc = grads[k].contiguous();
    AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, p.scalar_type(), "adam_step", [&] {
      using T = scalar_t;
      hipLaunchKernelGGL((adam_kernel<T>), dim3(blocks), dim3(kBlock), 0, stream.stream(), (T*)p.data_ptr(),
                         (const T*)gc.data_ptr(), ms[k].data_ptr<float>(), vs[k].data_ptr<float>(), n, (float)lr,
                         (float)b1, (float)b2, (float)eps, (float)wd, (float)bc1, (float)bc2);
    });
  }
}

// ---------------------------------------------------------------------------
// fused unimix-categorical straight-through head
// rows of K logits: p = (1-u)*softmax(L) + u/K ; m = log p ;
// sample = onehot(argmax(m + gumbel)) (or argmax(m) when !SAMPLE).
// One wave per row (K <= 64 lanes, looped above that).
// ---------------------------------------------------------------------------

template <typename T, bool SAMPLE>
__global__ void cat_st_fwd_kernel(const T* __restrict__ raw, const float* __restrict__ urand,
                                  float* __restrict__ m_out, T* __restrict__ onehot, float* __restrict__ s_out,
                                  long nrows, int K, float unimix, long ohs, long spb,
                                  T* __restrict__ onehot2, long oh2s) {
  const int lane = threadIdx.x & 63;
  const long row = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= nrows) return;
  const T* L = raw + row * (long)K;
  float* mr = m_out + row * (long)K;
  float* sr = s_out + row * (long)K;
  // onehot may be a [B, S*K] row-strided slice: row = b*spb + s maps to
  // b*ohs + s*K (ohs = S*K, spb = S reproduces the contiguous layout)
  T* oh = onehot + (row / spb) * ohs + (row % spb) * (long)K;
  T* oh2 = onehot2 ? onehot2 + (row / spb) * oh2s + (row % spb) * (long)K : nullptr;
  // row max
  float lmax = -1e30f;
  for (int j = lane; j < K; j += 64) lmax = fmaxf(lmax, ld(L, j));
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) lmax = fmaxf(lmax, __shfl_xor(lmax, off, 64));
  // exp + sum
  float lsum = 0.f;
  for (int j = lane; j < K; j += 64) lsum += texp<T>(ld(L, j) - lmax);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) lsum += __shfl_xor(lsum, off, 64);
  const float inv = 1.f / lsum;
  // p, m, gumbel-argmax
  float best = -1e30f;
  int best_j = 0;
  for (int j = lane; j < K; j += 64) {
    float s = texp<T>(ld(L, j) - lmax) * inv;
    float p = (1.f - unimix) * s + unimix / K;
    float m = tlog<T>(p);
    sr[j] = s;
    mr[j] = m;
    float score = m;
    if (SAMPLE) {
      float u = urand[row * (long)K + j];
      float t = fmaxf(-tlog<T>(fmaxf(u, 1e-20f)), 1e-20f);
      score += -tlog<T>(t);  // Gumbel(0,1) noise
    }
    if (score > best) {
      best = score;
      best_j = j;
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ob = __shfl_xor(best, off, 64);
    int oj = __shfl_xor(best_j, off, 64);
    if (ob > best || (ob == best && oj < best_j)) {
      best = ob;
      best_j = oj;
    }
  }
  for (int j = lane; j < K; j += 64) {
    const float z = j == best_j ? 1.f : 0.f;
    st(oh, j, z);
    if (oh2) st(oh2, j, z);
  }
}

// cat_st forward fused with the NEXT scan step's reset-masked input
// assembly: the z' block of x_{t+1} is the one-hot this kernel just sampled
// (masked in-register, no extra read), each (b,s) wave also writes its
// H/S-slice of the h' block of hu_{t+1} from h_t, and the s==S-1 wave adds
// the masked-action tail — removing the standalone scan_resets_fwd launch
// from steps 1..T-1 (~5 us/step on the DV3-S bench).
template <typename T>
__global__ void cat_st_resets_fwd_kernel(
    const T* __restrict__ raw, const float* __restrict__ urand, float* __restrict__ m_out,
    T* __restrict__ onehot, float* __restrict__ s_out, long nrows, int K, float unimix,
    const T* __restrict__ iz, const T* __restrict__ h_cur, const T* __restrict__ ih,
    const T* __restrict__ act, const T* __restrict__ f, T* __restrict__ x_next, long xs,
    T* __restrict__ hu_next, long hus, int S, int A, int H) {
  const int lane = threadIdx.x & 63;
  const long row = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= nrows) return;
  const long b = row / S;
  const int s = (int)(row - b * S);
  const T* L = raw + row * (long)K;
  float* mr = m_out + row * (long)K;
  float* sr = s_out + row * (long)K;
  T* oh = onehot + row * (long)K;
  const float fb = ld(f, b);
  const int SK = S * K;
  float lmax = -1e30f;
  for (int j = lane; j < K; j += 64) lmax = fmaxf(lmax, ld(L, j));
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) lmax = fmaxf(lmax, __shfl_xor(lmax, off, 64));
  float lsum = 0.f;
  for (int j = lane; j < K; j += 64) lsum += texp<T>(ld(L, j) - lmax);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) lsum += __shfl_xor(lsum, off, 64);
  const float inv = 1.f / lsum;
  float best = -1e30f;
  int best_j = 0;
  for (int j = lane; j < K; j += 64) {
    float sv = texp<T>(ld(L, j) - lmax) * inv;
    float p = (1.f - unimix) * sv + unimix / K;
    float m = tlog<T>(p);
    sr[j] = sv;
    mr[j] = m;
    float u = urand[row * (long)K + j];
    float t = fmaxf(-tlog<T>(fmaxf(u, 1e-20f)), 1e-20f);
    float score = m - tlog<T>(t);
    if (score > best) {
      best = score;
      best_j = j;
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ob = __shfl_xor(best, off, 64);
    int oj = __shfl_xor(best_j, off, 64);
    if (ob > best || (ob == best && oj < best_j)) {
      best = ob;
      best_j = oj;
    }
  }
  for (int j = lane; j < K; j += 64) {
    const float z = j == best_j ? 1.f : 0.f;
    st(oh, j, z);
    st(x_next, b * xs + s * K + j, (1.f - fb) * z + fb * ld(iz, b * (long)SK + s * K + j));
  }
  if (s == S - 1)
    for (int j = lane; j < A; j += 64) st(x_next, b * xs + SK + j, (1.f - fb) * ld(act, b * (long)A + j));
  const int HS = H / S;  // caller guarantees H % S == 0
  for (int j = lane; j < HS; j += 64) {
    const long c = (long)s * HS + j;
    st(hu_next, b * hus + c, (1.f - fb) * ld(h_cur, b * (long)H + c) + fb * ld(ih, b * (long)H + c));
  }
}

void cat_st_resets_fwd(const torch::Tensor& raw, const torch::Tensor& urand, double unimix,
                       torch::Tensor m, torch::Tensor onehot, torch::Tensor s_out,
                       const torch::Tensor& iz, const torch::Tensor& h_cur, const torch::Tensor& ih,
                       const torch::Tensor& act, const torch::Tensor& f, torch::Tensor x_next,
                       torch::Tensor hu_next) {
  CHECK_IN(raw);
  TORCH_CHECK(m.is_contiguous() && s_out.is_contiguous() && onehot.is_contiguous(), "cat_st_resets outputs");
  int K = (int)raw.size(-1);
  int S = (int)raw.size(-2);
  long nrows = raw.numel() / K;
  int A = (int)act.size(1);
  int H = (int)ih.size(1);
  TORCH_CHECK(H % S == 0, "cat_st_resets_fwd requires H % S == 0");
  TORCH_CHECK(S * K == (int)iz.size(1), "iz must be [B, S*K]");
  const int rows_per_block = kBlock / 64;
  int blocks = (int)((nrows + rows_per_block - 1) / rows_per_block);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, raw.scalar_type(), "cat_st_resets_fwd", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL((cat_st_resets_fwd_kernel<T>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       (const T*)raw.data_ptr(), urand.data_ptr<float>(), m.data_ptr<float>(),
                       (T*)onehot.data_ptr(), s_out.data_ptr<float>(), nrows, K, (float)unimix,
                       (const T*)iz.data_ptr(), (const T*)h_cur.data_ptr(), (const T*)ih.data_ptr(),
                       (const T*)act.data_ptr(), (const T*)f.data_ptr(), (T*)x_next.data_ptr(),
                       x_next.stride(0), (T*)hu_next.data_ptr(), hu_next.stride(0), S, A, H);
  });
}

// backward: given gm (grad wrt m=log p) and gon (grad wrt the ST sample whose
// gradient path is p), produce grad wrt raw logits.
// g_L_j = (1-unimix) * s_j * [ (gm_j/p_j + gon_j) - sum_i (gm_i/p_i + gon_i) * s_i ]
template <typename T>
__global__ void cat_st_bwd_kernel(const float* __restrict__ gm, const T* __restrict__ gon,
                                  const T* __restrict__ gon2, const float* __restrict__ s_saved,
                                  T* __restrict__ graw, long nrows, int K, float unimix) {
  const int lane = threadIdx.x & 63;
  const long row = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= nrows) return;
  const float* gmr = gm + row * (long)K;
  const T* gor = gon + row * (long)K;
  const T* gor2 = gon2 ? gon2 + row * (long)K : nullptr;
  const float* sr = s_saved + row * (long)K;
  T* gr = graw + row * (long)K;
  float acc = 0.f;
  for (int j = lane; j < K; j += 64) {
    float sj = sr[j];
    float pj = (1.f - unimix) * sj + unimix / K;
    float t = gmr[j] / pj + ld(gor, j) + (gor2 ? ld(gor2, j) : 0.f);
    acc += t * sj;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_xor(acc, off, 64);
  for (int j = lane; j < K; j += 64) {
    float sj = sr[j];
    float pj = (1.f - unimix) * sj + unimix / K;
    float t = gmr[j] / pj + ld(gor, j) + (gor2 ? ld(gor2, j) : 0.f);
    st(gr, j, (1.f - unimix) * sj * (t - acc));
  }
}

// cat_st backward for step t fused with scan_resets_bwd of step t+1: the
// z-carry entering this step's ST backward is (1-f_{t+1}) * gx_{t+1} — computed
// in-register instead of via a standalone launch; the same kernel emits the
// h-carry for this step's GRU backward, the action grad of t+1, and the
// masked init-state accumulator updates.
template <typename T>
__global__ void cat_st_resets_bwd_kernel(
    const float* __restrict__ gm, const T* __restrict__ gon, const float* __restrict__ s_saved,
    T* __restrict__ graw, long nrows, int K, float unimix,
    const T* __restrict__ ghu, long ghus, const T* __restrict__ ghp,
    const T* __restrict__ gx, long gxs, const T* __restrict__ f,
    T* __restrict__ gh_carry, T* __restrict__ ga, float* __restrict__ gih_acc,
    float* __restrict__ giz_acc, int S, int A, int H) {
  const int lane = threadIdx.x & 63;
  const long row = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= nrows) return;
  const long b = row / S;
  const int s = (int)(row - b * S);
  const int SK = S * K;
  const float fb = ld(f, b);
  const float* gmr = gm + row * (long)K;
  const T* gor = gon + row * (long)K;
  const float* sr = s_saved + row * (long)K;
  T* gr = graw + row * (long)K;
  float acc = 0.f;
  for (int j = lane; j < K; j += 64) {
    float sj = sr[j];
    float pj = (1.f - unimix) * sj + unimix / K;
    float gxv = ld(gx, b * gxs + s * K + j);
    giz_acc[b * (long)SK + s * K + j] += fb * gxv;
    float t = gmr[j] / pj + ld(gor, j) + (1.f - fb) * gxv;
    acc += t * sj;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_xor(acc, off, 64);
  for (int j = lane; j < K; j += 64) {
    float sj = sr[j];
    float pj = (1.f - unimix) * sj + unimix / K;
    float gxv = ld(gx, b * gxs + s * K + j);
    float t = gmr[j] / pj + ld(gor, j) + (1.f - fb) * gxv;
    st(gr, j, (1.f - unimix) * sj * (t - acc));
  }
  if (s == S - 1)
    for (int j = lane; j < A; j += 64) st(ga, b * (long)A + j, (1.f - fb) * ld(gx, b * gxs + SK + j));
  const int HS = H / S;  // caller guarantees H % S == 0
  for (int j = lane; j < HS; j += 64) {
    const long c = (long)s * HS + j;
    float g = ld(ghu, b * ghus + c) + ld(ghp, b * (long)H + c);
    st(gh_carry, b * (long)H + c, (1.f - fb) * g);
    gih_acc[b * (long)H + c] += fb * g;
  }
}

void cat_st_resets_bwd(const torch::Tensor& gm, const torch::Tensor& gon, const torch::Tensor& s_saved,
                       double unimix, torch::Tensor graw, const torch::Tensor& ghu,
                       const torch::Tensor& ghp, const torch::Tensor& gx, const torch::Tensor& f,
                       torch::Tensor gh_carry, torch::Tensor ga, torch::Tensor gih_acc,
                       torch::Tensor giz_acc) {
  CHECK_IN(gm);
  int K = (int)gm.size(-1);
  int S = (int)gm.size(-2);
  long nrows = gm.numel() / K;
  int A = (int)ga.size(1);
  int H = (int)gh_carry.size(1);
  TORCH_CHECK(H % S == 0, "cat_st_resets_bwd requires H % S == 0");
  const int rows_per_block = kBlock / 64;
  int blocks = (int)((nrows + rows_per_block - 1) / rows_per_block);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto gonc = gon.contiguous();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, graw.scalar_type(), "cat_st_resets_bwd", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL((cat_st_resets_bwd_kernel<T>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       gm.data_ptr<float>(), (const T*)gonc.data_ptr(), s_saved.data_ptr<float>(),
                       (T*)graw.data_ptr(), nrows, K, (float)unimix, (const T*)ghu.data_ptr(),
                       ghu.stride(0), (const T*)ghp.data_ptr(), (const T*)gx.data_ptr(), gx.stride(0),
                       (const T*)f.data_ptr(), (T*)gh_carry.data_ptr(), (T*)ga.data_ptr(),
                       gih_acc.data_ptr<float>(), giz_acc.data_ptr<float>(), S, A, H);
  });
}

std::vector<torch::Tensor> cat_st_fwd(const torch::Tensor& raw, const c10::optional<torch::Tensor>& urand,
                                      double unimix, bool sample) {
  CHECK_IN(raw);
  int K = (int)raw.size(-1);
  long nrows = raw.numel() / K;
  auto m = torch::empty(raw.sizes(), raw.options().dtype(at::kFloat));
  auto s = torch::empty(raw.sizes(), raw.options().dtype(at::kFloat));
  auto onehot = torch::empty_like(raw);
  const int rows_per_block = kBlock / 64;
  int blocks = (int)((nrows + rows_per_block - 1) / rows_per_block);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, raw.scalar_type(), "cat_st_fwd", [&] {
    using T = scalar_t;
    const float* up = urand.has_value() ? urand->data_ptr<float>() : nullptr;
    if (sample)
      hipLaunchKernelGGL((cat_st_fwd_kernel<T, true>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                         (const T*)raw.data_ptr(), up, m.data_ptr<float>(), (T*)onehot.data_ptr(),
                         s.data_ptr<float>(), nrows, K, (float)unimix, (long)K, 1, (T*)nullptr, 0L);
    else
      hipLaunchKernelGGL((cat_st_fwd_kernel<T, false>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                         (const T*)raw.data_ptr(), up, m.data_ptr<float>(), (T*)onehot.data_ptr(),
                         s.data_ptr<float>(), nrows, K, (float)unimix, (long)K, 1, (T*)nullptr, 0L);
  });
  return {m, onehot, s};
}

// Out-variant for the fused scan forward: m/onehot/s are caller-provided
// slices of stacked [T, ...] buffers (all contiguous); onehot is written in
// the compute dtype directly (no separate cast kernel).
void cat_st_fwd_o(const torch::Tensor& raw, const torch::Tensor& urand, double unimix, torch::Tensor m,
                  torch::Tensor onehot, torch::Tensor s,
                  const c10::optional<torch::Tensor>& onehot2 = c10::nullopt) {
  CHECK_IN(raw);
  TORCH_CHECK(m.is_contiguous() && s.is_contiguous(), "cat_st_fwd_o outputs");
  TORCH_CHECK(onehot.scalar_type() == raw.scalar_type(), "onehot dtype must match raw");
  int K = (int)raw.size(-1);
  long nrows = raw.numel() / K;
  // onehot is either contiguous (any shape) or a 2-D [B, S*K] row-strided
  // slice of a stacked buffer (fast imagination writes z straight into the
  // trajectory tensor)
  long ohs = (long)K, spb = 1;
  if (!onehot.is_contiguous()) {
    TORCH_CHECK(onehot.dim() == 2 && onehot.stride(1) == 1 && nrows % onehot.size(0) == 0,
                "cat_st_fwd_o: onehot must be contiguous or a row-strided 2-D slice");
    spb = nrows / onehot.size(0);
    ohs = onehot.stride(0);
  }
  const int rows_per_block = kBlock / 64;
  int blocks = (int)((nrows + rows_per_block - 1) / rows_per_block);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, raw.scalar_type(), "cat_st_fwd_o", [&] {
    using T = scalar_t;
    T* oh2 = nullptr;
    long oh2s = 0;
    if (onehot2.has_value()) {
      // same [rows-of-onehot, spb*K] geometry, its own row stride
      TORCH_CHECK(onehot2->dim() == 2 && onehot2->stride(1) == 1 &&
                      onehot2->scalar_type() == raw.scalar_type(),
                  "cat_st_fwd_o: onehot2 must be a row-strided 2-D slice");
      oh2 = (T*)onehot2->data_ptr();
      oh2s = onehot2->stride(0);
    }
    hipLaunchKernelGGL((cat_st_fwd_kernel<T, true>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       (const T*)raw.data_ptr(), urand.data_ptr<float>(), m.data_ptr<float>(),
                       (T*)onehot.data_ptr(), s.data_ptr<float>(), nrows, K, (float)unimix, ohs, spb,
                       oh2, oh2s);
  });
}

torch::Tensor cat_st_bwd(const torch::Tensor& gm, const torch::Tensor& gon, const torch::Tensor& s,
                         double unimix) {
  CHECK_IN(gm);
  int K = (int)gm.size(-1);
  long nrows = gm.numel() / K;
  auto graw = torch::empty(gm.sizes(), gm.options().dtype(gon.scalar_type()));
  const int rows_per_block = kBlock / 64;
  int blocks = (int)((nrows + rows_per_block - 1) / rows_per_block);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto gonc = gon.contiguous();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, graw.scalar_type(), "cat_st_bwd", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL((cat_st_bwd_kernel<T>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       gm.data_ptr<float>(), (const T*)gonc.data_ptr(), nullptr, s.data_ptr<float>(),
                       (T*)graw.data_ptr(), nrows, K, (float)unimix);
  });
  return graw;
}

// Out-variant for the fused scan backward: graw is a caller-provided slice;
// gon2 (optional, e.g. the step carry) is summed with gon inside the kernel.
void cat_st_bwd_o(const torch::Tensor& gm, const torch::Tensor& gon, const c10::optional<torch::Tensor>& gon2,
                  const torch::Tensor& s, double unimix, torch::Tensor graw) {
  CHECK_IN(gm);
  CHECK_IN(gon);
  TORCH_CHECK(graw.is_contiguous() && graw.scalar_type() == gon.scalar_type(), "cat_st_bwd_o output");
  int K = (int)gm.size(-1);
  long nrows = gm.numel() / K;
  const int rows_per_block = kBlock / 64;
  int blocks = (int)((nrows + rows_per_block - 1) / rows_per_block);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, graw.scalar_type(), "cat_st_bwd_o", [&] {
    using T = scalar_t;
    const T* g2 = gon2.has_value() ? (const T*)gon2->data_ptr() : nullptr;
    hipLaunchKernelGGL((cat_st_bwd_kernel<T>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       gm.data_ptr<float>(), (const T*)gon.data_ptr(), g2, s.data_ptr<float>(),
                       (T*)graw.data_ptr(), nrows, K, (float)unimix);
  });
}

// ---------------------------------------------------------------------------
// masked lerp (episode-reset masking in the RSSM scan):
//   y[b, j] = (1 - f[b]) * x[b, j] + f[b] * init[b, j]      (init optional)
// backward: gx = (1-f) * g ; ginit = f * g
// ---------------------------------------------------------------------------

template <typename T, bool HAS_INIT>
__global__ void masked_lerp_fwd_kernel(const T* __restrict__ x, const T* __restrict__ init,
                                       const T* __restrict__ f, T* __restrict__ y, long rows, int cols) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < rows * (long)cols;
       i += (long)gridDim.x * blockDim.x) {
    const long b = i / cols;
    float fb = ld(f, b);
    float v = (1.f - fb) * ld(x, i);
    if (HAS_INIT) v += fb * ld(init, i);
    st(y, i, v);
  }
}

template <typename T, bool HAS_INIT>
__global__ void masked_lerp_bwd_kernel(const T* __restrict__ g, const T* __restrict__ f, T* __restrict__ gx,
                                       T* __restrict__ ginit, long rows, int cols) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < rows * (long)cols;
       i += (long)gridDim.x * blockDim.x) {
    const long b = i / cols;
    float fb = ld(f, b);
    float gv = ld(g, i);
    st(gx, i, (1.f - fb) * gv);
    if (HAS_INIT) st(ginit, i, fb * gv);
  }
}

torch::Tensor masked_lerp_fwd(const torch::Tensor& x, const c10::optional<torch::Tensor>& init,
                              const torch::Tensor& f) {
  CHECK_IN(x);
  auto y = torch::empty_like(x);
  int cols = (int)x.size(-1);
  long rows = x.numel() / cols;
  int blocks = (int)std::min((x.numel() + kBlock - 1) / kBlock, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "masked_lerp_fwd", [&] {
    using T = scalar_t;
    if (init.has_value())
      hipLaunchKernelGGL((masked_lerp_fwd_kernel<T, true>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                         (const T*)x.data_ptr(), (const T*)init->data_ptr(), (const T*)f.data_ptr(),
                         (T*)y.data_ptr(), rows, cols);
    else
      hipLaunchKernelGGL((masked_lerp_fwd_kernel<T, false>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                         (const T*)x.data_ptr(), nullptr, (const T*)f.data_ptr(), (T*)y.data_ptr(), rows, cols);
  });
  return y;
}

std::vector<torch::Tensor> masked_lerp_bwd(const torch::Tensor& g, const torch::Tensor& f, bool has_init) {
  CHECK_IN(g);
  auto gx = torch::empty_like(g);
  auto ginit = has_init ? torch::empty_like(g) : torch::empty({0}, g.options());
  int cols = (int)g.size(-1);
  long rows = g.numel() / cols;
  int blocks = (int)std::min((g.numel() + kBlock - 1) / kBlock, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, g.scalar_type(), "masked_lerp_bwd", [&] {
    using T = scalar_t;
    if (has_init)
      hipLaunchKernelGGL((masked_lerp_bwd_kernel<T, true>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                         (const T*)g.data_ptr(), (const T*)f.data_ptr(), (T*)gx.data_ptr(), (T*)ginit.data_ptr(),
                         rows, cols);
    else
      hipLaunchKernelGGL((masked_lerp_bwd_kernel<T, false>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                         (const T*)g.data_ptr(), (const T*)f.data_ptr(), (T*)gx.data_ptr(), nullptr, rows, cols);
  });
  return {gx, ginit};
}

__global__ void step_inc_kernel(float* step_t) {
  if (threadIdx.x == 0 && blockIdx.x == 0) step_t[0] += 1.0f;
}

// Device-side bias correction (bc = 1 - beta^step read from a device scalar),
// so the whole optimizer step is hipGraph-replayable with correct step counts.
template <typename T>
__global__ void adam_dev_kernel(T* __restrict__ p, const T* __restrict__ g, float* __restrict__ m,
                                float* __restrict__ v, const float* __restrict__ step_t, long n, float lr, float b1,
                                float b2, float eps, float wd) {
  const float step = step_t[0];
  const float bc1 = 1.f - powf(b1, step);
  const float bc2 = 1.f - powf(b2, step);
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n; i += (long)gridDim.x * blockDim.x) {
    float gf = ld(g, i);
    float pf = ld(p, i);
    if (wd != 0.f) gf += wd * pf;
    float mi = b1 * m[i] + (1.f - b1) * gf;
    float vi = b2 * v[i] + (1.f - b2) * gf * gf;
    m[i] = mi;
    v[i] = vi;
    float upd = (mi / bc1) / (sqrtf(vi / bc2) + eps);
    st(p, i, pf - lr * upd);
  }
}

// Multi-tensor Adam over a prebuilt chunk table: ptrs [K,4] holds
// (p, g, m, v) addresses per tensor, sizes [K] the element counts, and
// ctid/coff [C] map each block to (tensor, start element).  One launch
// updates every tensor of the optimizer; when zero_grad is set the gradient
// is zeroed in place after being consumed, replacing the per-parameter
// zero_grad fills of the training loop.
constexpr long kAdamChunk = 4096;

template <typename T>
__global__ void adam_mt_kernel(const long* __restrict__ ptrs, const long* __restrict__ sizes,
                               const int* __restrict__ ctid, const long* __restrict__ coff,
                               const float* __restrict__ step_t, long C, float lr, float b1, float b2, float eps,
                               float wd, int zero_grad) {
  const long c = blockIdx.x;
  if (c >= C) return;
  const int k = ctid[c];
  const long off = coff[c];
  const long n = sizes[k];
  T* p = (T*)ptrs[4 * k];
  T* g = (T*)ptrs[4 * k + 1];
  float* m = (float*)ptrs[4 * k + 2];
  float* v = (float*)ptrs[4 * k + 3];
  const float step = step_t[0];
  const float bc1 = 1.f - powf(b1, step);
  const float bc2 = 1.f - powf(b2, step);
  const long end = (off + kAdamChunk < n) ? off + kAdamChunk : n;
  for (long i = off + threadIdx.x; i < end; i += blockDim.x) {
    float gf = ld(g, i);
    float pf = ld(p, i);
    if (wd != 0.f) gf += wd * pf;
    float mi = b1 * m[i] + (1.f - b1) * gf;
    float vi = b2 * v[i] + (1.f - b2) * gf * gf;
    m[i] = mi;
    v[i] = vi;
    float upd = (mi / bc1) / (sqrtf(vi / bc2) + eps);
    st(p, i, pf - lr * upd);
    if (zero_grad) st(g, i, 0.f);
  }
}

void adam_step_mt(const torch::Tensor& ptrs, const torch::Tensor& sizes, const torch::Tensor& ctid,
                  const torch::Tensor& coff, torch::Tensor step_t, const torch::Tensor& dtype_like,
                  bool inc_step, double lr, double b1, double b2, double eps, double wd, bool zero_grad) {
  auto stream = at::cuda::getCurrentCUDAStream();
  if (inc_step)
    hipLaunchKernelGGL(step_inc_kernel, dim3(1), dim3(64), 0, stream.stream(), step_t.data_ptr<float>());
  long C = ctid.numel();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, dtype_like.scalar_type(), "adam_step_mt", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL((adam_mt_kernel<T>), dim3((int)C), dim3(kBlock), 0, stream.stream(),
                       ptrs.data_ptr<long>(), sizes.data_ptr<long>(), ctid.data_ptr<int>(),
                       coff.data_ptr<long>(), step_t.data_ptr<float>(), C, (float)lr, (float)b1, (float)b2,
                       (float)eps, (float)wd, zero_grad ? 1 : 0);
  });
}

// TF-style RMSprop, one launch for the whole optimizer (same chunk-table
// scheme as adam_mt): v <- rho v + (1-rho) g^2; update g / sqrt(v + eps)
// (eps INSIDE the sqrt); optional momentum and centered variants
// (reference sheeprl/optim/rmsprop_tf.py:63-156).  ptrs rows:
// [p, g, square_avg, momentum_buf, grad_avg] (unused slots 0).
template <typename T>
__global__ void rmsprop_mt_kernel(const long* __restrict__ ptrs, const long* __restrict__ sizes,
                                  const int* __restrict__ ctid, const long* __restrict__ coff, long C,
                                  float lr, float alpha, float eps, float wd, float momentum,
                                  int centered, int zero_grad) {
  const long c = blockIdx.x;
  if (c >= C) return;
  const int k = ctid[c];
  const long off = coff[c];
  const long n = sizes[k];
  T* p = (T*)ptrs[5 * k];
  T* g = (T*)ptrs[5 * k + 1];
  float* sq = (float*)ptrs[5 * k + 2];
  float* mb = (float*)ptrs[5 * k + 3];
  float* ga = (float*)ptrs[5 * k + 4];
  const long end = (off + kAdamChunk < n) ? off + kAdamChunk : n;
  for (long i = off + threadIdx.x; i < end; i += blockDim.x) {
    float gf = ld(g, i);
    float pf = ld(p, i);
    if (wd != 0.f) gf += wd * pf;
    float s = sq[i] + (1.f - alpha) * (gf * gf - sq[i]);
    sq[i] = s;
    float avg;
    if (centered) {
      float a = ga[i] + (1.f - alpha) * (gf - ga[i]);
      ga[i] = a;
      avg = sqrtf(s - a * a + eps);  // reference does not clamp (rmsprop_tf.py:205)
    } else {
      avg = sqrtf(s + eps);
    }
    float upd;
    if (momentum > 0.f) {
      float b = momentum * mb[i] + gf / avg;
      mb[i] = b;
      upd = b;
    } else {
      upd = gf / avg;
    }
    st(p, i, pf - lr * upd);
    if (zero_grad) st(g, i, 0.f);
  }
}

void rmsprop_step_mt(const torch::Tensor& ptrs, const torch::Tensor& sizes, const torch::Tensor& ctid,
                     const torch::Tensor& coff, const torch::Tensor& proto, double lr, double alpha,
                     double eps, double wd, double momentum, bool centered, bool zero_grad) {
  long C = ctid.numel();
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, proto.scalar_type(), "rmsprop_mt", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL((rmsprop_mt_kernel<T>), dim3((int)C), dim3(kBlock), 0, stream.stream(),
                       ptrs.data_ptr<long>(), sizes.data_ptr<long>(), ctid.data_ptr<int>(),
                       coff.data_ptr<long>(), C, (float)lr, (float)alpha, (float)eps, (float)wd,
                       (float)momentum, centered ? 1 : 0, zero_grad ? 1 : 0);
  });
}

// Fused gradient clipping by global norm (torch.nn.utils.clip_grad_norm_
// semantics: coef = max_norm / (norm + 1e-6), clamped to 1) over the same
// flat chunk table as the optimizers; ptrs rows here are just [g].  Two
// launches replace the ~10-launch foreach path: (1) block-sum + one
// atomicAdd/block of the squared norm into out[0] (out[1] pre-zeroed holds
// the step's generation; out zeroed host-side / by a fill before), (2) a
// second kernel — the launch boundary orders it after ALL norm blocks —
// scales every grad in place and lets block 0 publish the norm to out[1].
template <typename T>
__global__ void gradsq_mt_kernel(const long* __restrict__ ptrs, const long* __restrict__ sizes,
                                 const int* __restrict__ ctid, const long* __restrict__ coff, long C,
                                 float* __restrict__ out) {
  __shared__ float lds[16];
  const long c = blockIdx.x;
  if (c >= C) return;
  const int k = ctid[c];
  const long off = coff[c];
  const long n = sizes[k];
  const T* g = (const T*)ptrs[k];
  const long end = (off + kAdamChunk < n) ? off + kAdamChunk : n;
  float s = 0.f;
  for (long i = off + threadIdx.x; i < end; i += blockDim.x) {
    float gf = ld(g, i);
    s += gf * gf;
  }
  s = block_sum(s, lds);
  if (threadIdx.x == 0) atomicAdd(out, s);
}

template <typename T>
__global__ void clip_apply_mt_kernel(const long* __restrict__ ptrs, const long* __restrict__ sizes,
                                     const int* __restrict__ ctid, const long* __restrict__ coff, long C,
                                     const float* __restrict__ sq, float* __restrict__ norm_out,
                                     float max_norm) {
  const long c = blockIdx.x;
  if (c >= C) return;
  const float norm = sqrtf(sq[0]);
  if (c == 0 && threadIdx.x == 0) norm_out[0] = norm;
  float coef = max_norm / (norm + 1e-6f);
  if (coef >= 1.f) return;  // no-op when already under the bound
  const int k = ctid[c];
  const long off = coff[c];
  const long n = sizes[k];
  T* g = (T*)ptrs[k];
  const long end = (off + kAdamChunk < n) ? off + kAdamChunk : n;
  for (long i = off + threadIdx.x; i < end; i += blockDim.x) st(g, i, ld(g, i) * coef);
}

void clip_grad_norm_mt(const torch::Tensor& ptrs, const torch::Tensor& sizes, const torch::Tensor& ctid,
                       const torch::Tensor& coff, torch::Tensor out, const torch::Tensor& proto,
                       double max_norm) {
  long C = ctid.numel();
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, proto.scalar_type(), "clip_grad_norm_mt", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL((gradsq_mt_kernel<T>), dim3((int)C), dim3(kBlock), 0, stream.stream(),
                       ptrs.data_ptr<long>(), sizes.data_ptr<long>(), ctid.data_ptr<int>(),
                       coff.data_ptr<long>(), C, out.data_ptr<float>());
    hipLaunchKernelGGL((clip_apply_mt_kernel<T>), dim3((int)C), dim3(kBlock), 0, stream.stream(),
                       ptrs.data_ptr<long>(), sizes.data_ptr<long>(), ctid.data_ptr<int>(),
                       coff.data_ptr<long>(), C, out.data_ptr<float>(), out.data_ptr<float>() + 1,
                       (float)max_norm);
  });
}

void adam_step_dev(std::vector<torch::Tensor> params, std::vector<torch::Tensor> grads,
                   std::vector<torch::Tensor> ms, std::vector<torch::Tensor> vs, torch::Tensor step_t, double lr,
                   double b1, double b2, double eps, double wd) {
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(step_inc_kernel, dim3(1), dim3(64), 0, stream.stream(), step_t.data_ptr<float>());
  for (size_t k = 0; k < params.size(); ++k) {
    auto& p = params[k];
    long n = p.numel();
    if (n == 0) continue;
    int blocks = (int)std::min((n + kBlock - 1) / kBlock, (long)2048);
    auto gc = grads[k].contiguous();
    AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, p.scalar_type(), "adam_step_dev", [&] {
      using T = scalar_t;
      hipLaunchKernelGGL((adam_dev_kernel<T>), dim3(blocks), dim3(kBlock), 0, stream.stream(), (T*)p.data_ptr(),
                         (const T*)gc.data_ptr(), ms[k].data_ptr<float>(), vs[k].data_ptr<float>(),
                         step_t.data_ptr<float>(), n, (float)lr, (float)b1, (float)b2, (float)eps, (float)wd);
    });
  }
}

template <typename T>
__global__ void ema_kernel(T* __restrict__ t, const T* __restrict__ s, long n, float tau) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n; i += (long)gridDim.x * blockDim.x) {
    st(t, i, (1.f - tau) * ld(t, i) + tau * ld(s, i));
  }
}

void ema_update(std::vector<torch::Tensor> tgts, std::vector<torch::Tensor> srcs, double tau) {
  auto stream = at::cuda::getCurrentCUDAStream();
  for (size_t k = 0; k < tgts.size(); ++k) {
    long n = tgts[k].numel();
    if (n == 0) continue;
    int blocks = (int)std::min((n + kBlock - 1) / kBlock, (long)2048);
    AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, tgts[k].scalar_type(), "ema_update", [&] {
      using T = scalar_t;
      hipLaunchKernelGGL((ema_kernel<T>), dim3(blocks), dim3(kBlock), 0, stream.stream(), (T*)tgts[k].data_ptr(),
                         (const T*)srcs[k].data_ptr(), n, (float)tau);
    });
  }
}

// ---------------------------------------------------------------------------
// uint8 obs -> float in [-0.5, 0.5]
// ---------------------------------------------------------------------------

__global__ void obs_norm_kernel(const unsigned char* __restrict__ x, float* __restrict__ y, long n) {
  const long n4 = n / 4;
  const uchar4* x4 = reinterpret_cast<const uchar4*>(x);
  float4* y4 = reinterpret_cast<float4*>(y);
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n4; i += (long)gridDim.x * blockDim.x) {
    uchar4 v = x4[i];
    y4[i] = make_float4(v.x / 255.f - 0.5f, v.y / 255.f - 0.5f, v.z / 255.f - 0.5f, v.w / 255.f - 0.5f);
  }
  for (long i = n4 * 4 + blockIdx.x * (long)blockDim.x + threadIdx.x; i < n; i += (long)gridDim.x * blockDim.x)
    y[i] = x[i] / 255.f - 0.5f;
}

torch::Tensor obs_norm(const torch::Tensor& x) {
  CHECK_IN(x);
  TORCH_CHECK(x.scalar_type() == at::kByte, "obs_norm expects uint8");
  auto y = torch::empty(x.sizes(), x.options().dtype(at::kFloat));
  long n = x.numel();
  int blocks = (int)std::min((n / 4 + kBlock - 1) / kBlock + 1, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(obs_norm_kernel, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                     (const unsigned char*)x.data_ptr(), y.data_ptr<float>(), n);
  return y;
}

}  // namespace

// ---------------------------------------------------------------------------
// channels-last per-channel sum (conv bias gradient).  torch's bf16
// column-reduce on a [N*H*W, C] channels-last grad was measured at 1.3 ms for
// the DV3 decoder's final deconv; this kernel is a coalesced grid-stride sum
// with one LDS image per block and one global atomicAdd per channel per block.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void chlast_sum_kernel(const T* __restrict__ g, float* __restrict__ out, long R, int C) {
  extern __shared__ __attribute__((aligned(16))) float acc[];  // [C]
  for (int j = threadIdx.x; j < C; j += blockDim.x) acc[j] = 0.f;
  __syncthreads();
  const long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = (long)gridDim.x * blockDim.x;
  const long teff = (total / C) * C;  // whole-row multiple so the stride partition is exact
  if (tid < teff) {
    const int col = (int)(tid % C);
    const long rstep = teff / C;
    float s = 0.f;
    for (long r = tid / C; r < R; r += rstep) s += ld(g, r * C + col);
    atomicAdd(&acc[col], s);
  }
  __syncthreads();
  for (int j = threadIdx.x; j < C; j += blockDim.x) atomicAdd(&out[j], acc[j]);
}

torch::Tensor chlast_bias_sum(const torch::Tensor& g, long C) {
  TORCH_CHECK(g.is_cuda() && g.numel() % C == 0, "chlast_bias_sum shape");
  long R = g.numel() / C;
  auto out = torch::zeros({C}, g.options().dtype(at::kFloat));
  int blocks = (int)std::min((R * C + kBlock - 1) / kBlock, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, g.scalar_type(), "chlast_bias_sum", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL((chlast_sum_kernel<T>), dim3(blocks), dim3(kBlock), C * sizeof(float), stream.stream(),
                       (const T*)g.data_ptr(), out.data_ptr<float>(), R, (int)C);
  });
  return out;
}

// ---------------------------------------------------------------------------
// fused two-hot log-prob (DV3 reward/critic heads): logsumexp + uniform-bin
// two-hot encoding of symlog(value) + cross-entropy in one kernel each way
// (replaces the ~12-launch clamp/searchsorted/scatter/logsumexp chain of
// TwoHotEncodingDistribution.log_prob, distribution.py:224-276).
// ---------------------------------------------------------------------------

__device__ __forceinline__ void twohot_idx(float x, float low, float high, int K, int& lo, int& hi,
                                           float& w_lo, float& w_hi) {
  const float step = (high - low) / (K - 1);
  float a = fabsf(x);
  float sx = (x >= 0.f ? 1.f : -1.f) * log1pf(a);  // symlog
  sx = fminf(fmaxf(sx, low), high);
  hi = (int)ceilf((sx - low) / step);
  hi = min(max(hi, 0), K - 1);
  lo = max(hi - 1, 0);
  const float lo_v = low + lo * step;
  const float hi_v = low + hi * step;
  const float denom = fmaxf(hi_v - lo_v, 1e-8f);
  w_hi = fminf(fmaxf((sx - lo_v) / denom, 0.f), 1.f);
  w_lo = 1.f - w_hi;
}

__global__ void twohot_lp_fwd_kernel(const float* __restrict__ logits, const float* __restrict__ value,
                                     float* __restrict__ out, float* __restrict__ lse_out, long N, int K,
                                     float low, float high) {
  const int lane = threadIdx.x & 63;
  const long row = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= N) return;
  const float* lr = logits + row * (long)K;
  float mx = -1e30f;
  for (int j = lane; j < K; j += 64) mx = fmaxf(mx, lr[j]);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) mx = fmaxf(mx, __shfl_xor(mx, off, 64));
  float se = 0.f;
  for (int j = lane; j < K; j += 64) se += expf(lr[j] - mx);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) se += __shfl_xor(se, off, 64);
  const float lse = mx + logf(se);
  if (lane == 0) {
    int lo, hi;
    float wl, wh;
    twohot_idx(value[row], low, high, K, lo, hi, wl, wh);
    out[row] = wl * (lr[lo] - lse) + wh * (lr[hi] - lse);
    lse_out[row] = lse;
  }
}

__global__ void twohot_lp_bwd_kernel(const float* __restrict__ g, const float* __restrict__ logits,
                                     const float* __restrict__ value, const float* __restrict__ lse,
                                     float* __restrict__ gl, long N, int K, float low, float high) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < N * (long)K;
       i += (long)gridDim.x * blockDim.x) {
    const long row = i / K;
    const int j = (int)(i - row * K);
    int lo, hi;
    float wl, wh;
    twohot_idx(value[row], low, high, K, lo, hi, wl, wh);
    const float tw = (j == lo ? wl : 0.f) + (j == hi ? wh : 0.f);
    gl[i] = g[row] * (tw - expf(logits[i] - lse[row]));
  }
}

std::vector<torch::Tensor> twohot_lp_fwd(const torch::Tensor& logits, const torch::Tensor& value, double low,
                                         double high) {
  CHECK_IN(logits);
  TORCH_CHECK(logits.scalar_type() == at::kFloat && value.scalar_type() == at::kFloat);
  int K = (int)logits.size(-1);
  long N = logits.numel() / K;
  auto out = torch::empty({N}, logits.options());
  auto lse = torch::empty({N}, logits.options());
  const int rpb = kBlock / 64;
  int blocks = (int)((N + rpb - 1) / rpb);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(twohot_lp_fwd_kernel, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                     logits.data_ptr<float>(), value.data_ptr<float>(), out.data_ptr<float>(),
                     lse.data_ptr<float>(), N, K, (float)low, (float)high);
  return {out, lse};
}

torch::Tensor twohot_lp_bwd(const torch::Tensor& g, const torch::Tensor& logits, const torch::Tensor& value,
                            const torch::Tensor& lse, double low, double high) {
  int K = (int)logits.size(-1);
  long N = logits.numel() / K;
  auto gl = torch::empty_like(logits);
  int blocks = (int)std::min((N * K + kBlock - 1) / kBlock, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(twohot_lp_bwd_kernel, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                     g.data_ptr<float>(), logits.data_ptr<float>(), value.data_ptr<float>(),
                     lse.data_ptr<float>(), gl.data_ptr<float>(), N, K, (float)low, (float)high);
  return gl;
}

// ---------------------------------------------------------------------------
// fused balanced categorical KL (DV3 world-model loss, loss.py:64-75): both
// KL(sg(post)||prior) and KL(post||sg(prior)) share the forward VALUE; one
// kernel computes the per-sample KL (summed over stoch groups), the backward
// kernel routes the two incoming grads to prior/post respectively.
// ---------------------------------------------------------------------------

__global__ void klbal_fwd_kernel(const float* __restrict__ post, const float* __restrict__ prior,
                                 float* __restrict__ out, float* __restrict__ kls, long NS, int S, int K) {
  const int lane = threadIdx.x & 63;
  const long row = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);  // (n, s) pair
  if (row >= NS) return;
  const float* pr = post + row * (long)K;
  const float* qr = prior + row * (long)K;
  float mp = -1e30f, mq = -1e30f;
  for (int j = lane; j < K; j += 64) {
    mp = fmaxf(mp, pr[j]);
    mq = fmaxf(mq, qr[j]);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    mp = fmaxf(mp, __shfl_xor(mp, off, 64));
    mq = fmaxf(mq, __shfl_xor(mq, off, 64));
  }
  float sp = 0.f, sq = 0.f;
  for (int j = lane; j < K; j += 64) {
    sp += expf(pr[j] - mp);
    sq += expf(qr[j] - mq);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    sp += __shfl_xor(sp, off, 64);
    sq += __shfl_xor(sq, off, 64);
  }
  const float lsep = mp + logf(sp), lseq = mq + logf(sq);
  float kl = 0.f;
  for (int j = lane; j < K; j += 64) {
    const float lp = pr[j] - lsep;
    const float lq = qr[j] - lseq;
    kl += expf(lp) * (lp - lq);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) kl += __shfl_xor(kl, off, 64);
  if (lane == 0) {
    kls[row] = kl;
    atomicAdd(&out[row / S], kl);
  }
}

__global__ void klbal_bwd_kernel(const float* __restrict__ g_dyn, const float* __restrict__ g_rep,
                                 const float* __restrict__ post, const float* __restrict__ prior,
                                 const float* __restrict__ kls, float* __restrict__ g_post,
                                 float* __restrict__ g_prior, long NS, int S, int K) {
  const int lane = threadIdx.x & 63;
  const long row = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= NS) return;
  const float* pr = post + row * (long)K;
  const float* qr = prior + row * (long)K;
  float mp = -1e30f, mq = -1e30f;
  for (int j = lane; j < K; j += 64) {
    mp = fmaxf(mp, pr[j]);
    mq = fmaxf(mq, qr[j]);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    mp = fmaxf(mp, __shfl_xor(mp, off, 64));
    mq = fmaxf(mq, __shfl_xor(mq, off, 64));
  }
  float sp = 0.f, sq = 0.f;
  for (int j = lane; j < K; j += 64) {
    sp += expf(pr[j] - mp);
    sq += expf(qr[j] - mq);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    sp += __shfl_xor(sp, off, 64);
    sq += __shfl_xor(sq, off, 64);
  }
  const float lsep = mp + logf(sp), lseq = mq + logf(sq);
  const float gd = g_dyn[row / S], gr2 = g_rep[row / S];
  const float klg = kls[row];
  for (int j = lane; j < K; j += 64) {
    const float lp = pr[j] - lsep;
    const float lq = qr[j] - lseq;
    const float p = expf(lp);
    const float q = expf(lq);
    g_prior[row * (long)K + j] = gd * (q - p);
    g_post[row * (long)K + j] = gr2 * p * ((lp - lq) - klg);
  }
}

std::vector<torch::Tensor> klbal_fwd(const torch::Tensor& post, const torch::Tensor& prior) {
  CHECK_IN(post);
  CHECK_IN(prior);
  TORCH_CHECK(post.scalar_type() == at::kFloat && post.dim() >= 2);
  int K = (int)post.size(-1);
  int S = (int)post.size(-2);
  long NS = post.numel() / K;
  auto out = torch::zeros({NS / S}, post.options());
  auto kls = torch::empty({NS}, post.options());
  const int rpb = kBlock / 64;
  int blocks = (int)((NS + rpb - 1) / rpb);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(klbal_fwd_kernel, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                     post.data_ptr<float>(), prior.data_ptr<float>(), out.data_ptr<float>(),
                     kls.data_ptr<float>(), NS, S, K);
  return {out, kls};
}

std::vector<torch::Tensor> klbal_bwd(const torch::Tensor& g_dyn, const torch::Tensor& g_rep,
                                     const torch::Tensor& post, const torch::Tensor& prior,
                                     const torch::Tensor& kls) {
  int K = (int)post.size(-1);
  int S = (int)post.size(-2);
  long NS = post.numel() / K;
  auto g_post = torch::empty_like(post);
  auto g_prior = torch::empty_like(prior);
  const int rpb = kBlock / 64;
  int blocks = (int)((NS + rpb - 1) / rpb);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(klbal_bwd_kernel, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                     g_dyn.contiguous().data_ptr<float>(), g_rep.contiguous().data_ptr<float>(),
                     post.data_ptr<float>(), prior.data_ptr<float>(), kls.data_ptr<float>(),
                     g_post.data_ptr<float>(), g_prior.data_ptr<float>(), NS, S, K);
  return {g_post, g_prior};
}

// ---------------------------------------------------------------------------
// fused per-step episode-reset kernels for the RSSM scan: the three
// masked-lerp launches of a step (z', a', h') collapse into ONE kernel
// (regions decoded from the flat index), writing straight into the strided
// column blocks of the stacked x_s / hu_s buffers.
// ---------------------------------------------------------------------------

template <typename T, bool T0>
__global__ void scan_resets_fwd_kernel(const T* __restrict__ z_prev, const T* __restrict__ iz,
                                       const T* __restrict__ h_prev, const T* __restrict__ ih,
                                       const T* __restrict__ actions, const T* __restrict__ f,
                                       T* __restrict__ x, long xs, T* __restrict__ hu, long hus, long B, int SK,
                                       int A, int H) {
  const long n = B * (long)(SK + A + H);
  const int cols = SK + A + H;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n; i += (long)gridDim.x * blockDim.x) {
    const long m = i / cols;
    const int c = (int)(i - m * cols);
    const float fb = ld(f, m);
    if (c < SK) {
      float zp = T0 ? 0.f : ld(z_prev, m * SK + c);
      st(x, m * xs + c, (1.f - fb) * zp + fb * ld(iz, m * SK + c));
    } else if (c < SK + A) {
      st(x, m * xs + c, (1.f - fb) * ld(actions, m * A + (c - SK)));
    } else {
      const int j = c - SK - A;
      float hp = T0 ? 0.f : ld(h_prev, m * H + j);
      st(hu, m * hus + j, (1.f - fb) * hp + fb * ld(ih, m * H + j));
    }
  }
}

void scan_resets_fwd(const torch::Tensor& z_prev, const torch::Tensor& iz, const torch::Tensor& h_prev,
                     const torch::Tensor& ih, const torch::Tensor& actions, const torch::Tensor& f,
                     torch::Tensor x, torch::Tensor hu, bool t0) {
  long B = actions.size(0);
  int A = (int)actions.size(1);
  int SK = (int)iz.size(1);
  int H = (int)ih.size(1);
  long n = B * (long)(SK + A + H);
  int blocks = (int)std::min((n + kBlock - 1) / kBlock, (long)1024);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, actions.scalar_type(), "scan_resets_fwd", [&] {
    using T = scalar_t;
    if (t0)
      hipLaunchKernelGGL((scan_resets_fwd_kernel<T, true>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                         (const T*)z_prev.data_ptr(), (const T*)iz.data_ptr(), (const T*)h_prev.data_ptr(),
                         (const T*)ih.data_ptr(), (const T*)actions.data_ptr(), (const T*)f.data_ptr(),
                         (T*)x.data_ptr(), x.stride(0), (T*)hu.data_ptr(), hu.stride(0), B, SK, A, H);
    else
      hipLaunchKernelGGL((scan_resets_fwd_kernel<T, false>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                         (const T*)z_prev.data_ptr(), (const T*)iz.data_ptr(), (const T*)h_prev.data_ptr(),
                         (const T*)ih.data_ptr(), (const T*)actions.data_ptr(), (const T*)f.data_ptr(),
                         (T*)x.data_ptr(), x.stride(0), (T*)hu.data_ptr(), hu.stride(0), B, SK, A, H);
  });
}

// backward of the fused resets: consumes gh_in (= ghu[:, :H] + ghp summed
// in-kernel) and gx (strided [z', a'] block), producing the carries, the
// action grad, and the fp32 init-state accumulators — one launch instead of
// three.
template <typename T>
__global__ void scan_resets_bwd_kernel(const T* __restrict__ ghu, long ghus, const T* __restrict__ ghp,
                                       const T* __restrict__ gx, long gxs, const T* __restrict__ f,
                                       T* __restrict__ gh_carry, T* __restrict__ gz_carry,
                                       T* __restrict__ ga, float* __restrict__ gih_acc,
                                       float* __restrict__ giz_acc, long B, int SK, int A, int H) {
  const long n = B * (long)(SK + A + H);
  const int cols = SK + A + H;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n; i += (long)gridDim.x * blockDim.x) {
    const long m = i / cols;
    const int c = (int)(i - m * cols);
    const float fb = ld(f, m);
    if (c < SK) {
      float g = ld(gx, m * gxs + c);
      st(gz_carry, m * SK + c, (1.f - fb) * g);
      giz_acc[m * SK + c] += fb * g;
    } else if (c < SK + A) {
      st(ga, m * A + (c - SK), (1.f - fb) * ld(gx, m * gxs + (c - 0)));
    } else {
      const int j = c - SK - A;
      float g = ld(ghu, m * ghus + j) + ld(ghp, m * H + j);
      st(gh_carry, m * H + j, (1.f - fb) * g);
      gih_acc[m * H + j] += fb * g;
    }
  }
}

void scan_resets_bwd(const torch::Tensor& ghu, const torch::Tensor& ghp, const torch::Tensor& gx,
                     const torch::Tensor& f, torch::Tensor gh_carry, torch::Tensor gz_carry, torch::Tensor ga,
                     torch::Tensor gih_acc, torch::Tensor giz_acc) {
  long B = ga.size(0);
  int A = (int)ga.size(1);
  int SK = (int)gz_carry.size(1);
  int H = (int)gh_carry.size(1);
  long n = B * (long)(SK + A + H);
  int blocks = (int)std::min((n + kBlock - 1) / kBlock, (long)1024);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, ga.scalar_type(), "scan_resets_bwd", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL((scan_resets_bwd_kernel<T>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       (const T*)ghu.data_ptr(), ghu.stride(0), (const T*)ghp.data_ptr(),
                       (const T*)gx.data_ptr(), gx.stride(0), (const T*)f.data_ptr(),
                       (T*)gh_carry.data_ptr(), (T*)gz_carry.data_ptr(), (T*)ga.data_ptr(),
                       gih_acc.data_ptr<float>(), giz_acc.data_ptr<float>(), B, SK, A, H);
  });
}

// ---------------------------------------------------------------------------
// persistent fused RSSM scan — building blocks
// ---------------------------------------------------------------------------
// MFMA fragment types for v_mfma_f32_16x16x32_bf16 (gfx950): 8 bf16 per lane
// for A/B (A: row=lane&15, k=(lane>>4)*8+e; B: col=lane&15, same k), 4 fp32
// accumulators per lane (C/D: col=lane&15, row=(lane>>4)*4+reg).
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// ---------------------------------------------------------------------------
// M=16 GEMM family for the RSSM scan steps (per the CDNA4 guide's M<=16
// recipe: operands straight to VGPRs, deep unroll, no LDS staging for the
// streamed weight).  A is the [B<=16, K] activation (row-strided, unpadded:
// the final K-chunk is guard-loaded), W is the torch [N, K] weight.
// ---------------------------------------------------------------------------

__device__ __forceinline__ bf16x8 g16_load8v(const __hip_bfloat16* base, long row, long stride, int k) {
  return *(const bf16x8*)(base + row * stride + k);
}

__device__ __forceinline__ bf16x8 g16_load8(const __hip_bfloat16* base, long row, long stride, int k, int K,
                                            int nrows, int r) {
  bf16x8 v;
  if (r >= nrows) {
#pragma unroll
    for (int e = 0; e < 8; ++e) v[e] = (__bf16)0.f;
    return v;
  }
  const __hip_bfloat16* p = base + row * stride + k;
  if (k + 8 <= K) return *(const bf16x8*)p;
#pragma unroll
  for (int e = 0; e < 8; ++e) v[e] = (k + e < K) ? (__bf16)__bfloat162float(p[e]) : (__bf16)0.f;
  return v;
}

// plain multi-workgroup variant: C[16, N] (strided rows) = A @ W^T; one
// 16-col tile per wave, grid = N/64 workgroups.
__global__ void __launch_bounds__(256) g16_plain_kernel(const __hip_bfloat16* __restrict__ A, long as_,
                                                        const __hip_bfloat16* __restrict__ W,
                                                        const __hip_bfloat16* __restrict__ bias,
                                                        __hip_bfloat16* __restrict__ C, long cs, int B, int N,
                                                        int K) {
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int ncol0 = (blockIdx.x * (int)(blockDim.x >> 6) + (threadIdx.x >> 6)) * 16;
  if (ncol0 >= N) return;
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int KU = K & ~31;
  const int arow_c = arow < B ? arow : B - 1;  // clamp: garbage rows unused
  // two-phase body: issue all 16 loads of a 256-deep K block, then run the 8
  // MFMAs — keeps 16 loads in flight per wave (the "late vmcnt" idiom for
  // M<=16 streamed-weight GEMMs, cdna_hip_programming.md §6)
  const int KU8 = KU & ~255;
  for (int k0 = 0; k0 < KU8; k0 += 256) {
    bf16x8 af[8], bf[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int k = k0 + u * 32 + kgrp * 8;
      af[u] = g16_load8v(A, arow_c, as_, k);
      bf[u] = g16_load8v(W, ncol0 + arow, K, k);
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[u], bf[u], acc, 0, 0, 0);
  }
#pragma unroll 4
  for (int k0 = KU8; k0 < KU; k0 += 32) {
    const int k = k0 + kgrp * 8;
    bf16x8 a = g16_load8v(A, arow_c, as_, k);
    bf16x8 b = g16_load8v(W, ncol0 + arow, K, k);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
  if (K & 31) {
    const int k = KU + kgrp * 8;
    bf16x8 a = g16_load8(A, arow, as_, k, K, B, arow);
    bf16x8 b = g16_load8(W, ncol0 + arow, K, k, K, N, ncol0 + arow);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
  const float bv = bias ? __bfloat162float(bias[ncol0 + arow]) : 0.f;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = kgrp * 4 + r;
    if (m < B) C[(long)m * cs + ncol0 + arow] = (__hip_bfloat16)(acc[r] + bv);
  }
}

// split-K variant for long-K scan GEMMs (K ~ 6k: the representation-model
// input): KS K-slices per 64-col tile accumulate into an fp32 scratch with
// device atomics; the LAST-arriving workgroup of each tile (ticket counter,
// one agent-scope acquire per tile — the in-launch split-K reduction of
// cdna_hip_programming.md §6) converts the tile to bf16 and resets the
// scratch/ticket for the next call.
__global__ void __launch_bounds__(256) g16_splitk_kernel(const __hip_bfloat16* __restrict__ A, long as_,
                                                         const __hip_bfloat16* __restrict__ W,
                                                         const __hip_bfloat16* __restrict__ bias,
                                                         float* __restrict__ scratch, int* __restrict__ tickets,
                                                         __hip_bfloat16* __restrict__ C, long cs, int B, int N,
                                                         int K, int KS) {
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int ntiles = N / 64;
  const int tile = blockIdx.x % ntiles;
  const int slice = blockIdx.x / ntiles;
  const int ncol0 = tile * 64 + (int)(threadIdx.x >> 6) * 16;
  const int klen32 = (((K + KS - 1) / KS + 31) / 32) * 32;
  const int kbeg = slice * klen32;
  const int kend = min(kbeg + klen32, K);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int arow_c = arow < B ? arow : B - 1;
  const int KU = kbeg + ((kend - kbeg) & ~31);
  const int KU8 = kbeg + ((KU - kbeg) & ~255);
  for (int k0 = kbeg; k0 < KU8; k0 += 256) {
    bf16x8 af[8], bf[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int k = k0 + u * 32 + kgrp * 8;
      af[u] = g16_load8v(A, arow_c, as_, k);
      bf[u] = g16_load8v(W, ncol0 + arow, K, k);
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[u], bf[u], acc, 0, 0, 0);
  }
#pragma unroll 4
  for (int k0 = KU8; k0 < KU; k0 += 32) {
    const int k = k0 + kgrp * 8;
    bf16x8 a = g16_load8v(A, arow_c, as_, k);
    bf16x8 b = g16_load8v(W, ncol0 + arow, K, k);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
  if (KU < kend) {
    const int k = KU + kgrp * 8;
    bf16x8 a = g16_load8(A, arow, as_, k, kend, B, arow);
    bf16x8 b = g16_load8(W, ncol0 + arow, K, k, kend, N, ncol0 + arow);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = kgrp * 4 + r;
    atomicAdd(&scratch[(long)m * N + ncol0 + arow], acc[r]);
  }
  // tile episode hand-off: last arriver converts + resets
  __shared__ int last;
  __syncthreads();
  if (threadIdx.x == 0) {
    int t = __hip_atomic_fetch_add(&tickets[tile], 1, __ATOMIC_ACQ_REL, __HIP_MEMORY_SCOPE_AGENT);
    last = (t == KS - 1) ? 1 : 0;
    if (last) __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  }
  __syncthreads();
  if (last) {
    for (int i = threadIdx.x; i < 16 * 64; i += blockDim.x) {
      const int m = i >> 6, j = tile * 64 + (i & 63);
      const float v = scratch[(long)m * N + j];
      if (m < B) C[(long)m * cs + j] = (__hip_bfloat16)(v + (bias ? __bfloat162float(bias[j]) : 0.f));
      scratch[(long)m * N + j] = 0.f;
    }
    if (threadIdx.x == 0) __hip_atomic_store(&tickets[tile], 0, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  }
}

void g16_splitk(const torch::Tensor& A, const torch::Tensor& W, const c10::optional<torch::Tensor>& bias,
                torch::Tensor scratch, torch::Tensor tickets, torch::Tensor C, long KS) {
  TORCH_CHECK(A.dim() == 2 && A.stride(1) == 1 && W.is_contiguous() && C.stride(1) == 1);
  TORCH_CHECK(A.size(0) <= 16 && W.size(0) % 64 == 0 && A.scalar_type() == at::kBFloat16);
  int B = (int)A.size(0), K = (int)A.size(1), N = (int)W.size(0);
  TORCH_CHECK(scratch.numel() >= 16 * (long)N && tickets.numel() >= N / 64);
  auto stream = at::cuda::getCurrentCUDAStream();
  const __hip_bfloat16* bp = bias.has_value() ? (const __hip_bfloat16*)bias->data_ptr() : nullptr;
  hipLaunchKernelGGL(g16_splitk_kernel, dim3((N / 64) * (int)KS), dim3(256), 0, stream.stream(),
                     (const __hip_bfloat16*)A.data_ptr(), A.stride(0), (const __hip_bfloat16*)W.data_ptr(), bp,
                     scratch.data_ptr<float>(), tickets.data_ptr<int>(), (__hip_bfloat16*)C.data_ptr(),
                     C.stride(0), B, N, K, (int)KS);
}

void g16_plain(const torch::Tensor& A, const torch::Tensor& W, const c10::optional<torch::Tensor>& bias,
               torch::Tensor C) {
  TORCH_CHECK(A.dim() == 2 && A.stride(1) == 1 && W.is_contiguous() && C.stride(1) == 1);
  TORCH_CHECK(A.size(0) <= 16 && W.size(0) % 64 == 0 && A.scalar_type() == at::kBFloat16);
  int B = (int)A.size(0), K = (int)A.size(1), N = (int)W.size(0);
  auto stream = at::cuda::getCurrentCUDAStream();
  const __hip_bfloat16* bp = bias.has_value() ? (const __hip_bfloat16*)bias->data_ptr() : nullptr;
  hipLaunchKernelGGL(g16_plain_kernel, dim3(N / 64), dim3(256), 0, stream.stream(),
                     (const __hip_bfloat16*)A.data_ptr(), A.stride(0), (const __hip_bfloat16*)W.data_ptr(), bp,
                     (__hip_bfloat16*)C.data_ptr(), C.stride(0), B, N, K);
}

// single-workgroup GEMM + rowwise LN+SiLU epilogue (the recurrent-model MLP
// step): writes the pre-LN GEMM result (saved for backward), mean/rstd, and
// the activated output into a strided slice.  N <= 512 (LDS image).
template <int TILES>
__global__ void __launch_bounds__(256) g16_ln_silu_kernel(const __hip_bfloat16* __restrict__ A, long as_,
                                                          const __hip_bfloat16* __restrict__ W,
                                                          const __hip_bfloat16* __restrict__ lnw,
                                                          const __hip_bfloat16* __restrict__ lnb,
                                                          __hip_bfloat16* __restrict__ G, long gs,
                                                          __hip_bfloat16* __restrict__ Y, long ys,
                                                          float* __restrict__ mean_out,
                                                          float* __restrict__ rstd_out, int B, int N, int K,
                                                          float eps) {
  extern __shared__ __attribute__((aligned(16))) float lds[];  // [16][N]
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int wv = threadIdx.x >> 6;
  f32x4 acc[TILES];
#pragma unroll
  for (int t = 0; t < TILES; ++t) acc[t] = (f32x4){0.f, 0.f, 0.f, 0.f};
  const int KU = K & ~31;
  const int arow_c = arow < B ? arow : B - 1;
#pragma unroll 4
  for (int k0 = 0; k0 < KU; k0 += 32) {
    const int k = k0 + kgrp * 8;
    bf16x8 a = g16_load8v(A, arow_c, as_, k);
#pragma unroll
    for (int t = 0; t < TILES; ++t) {
      const int ncol0 = (t * 4 + wv) * 16;
      bf16x8 b = g16_load8v(W, ncol0 + arow, K, k);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[t], 0, 0, 0);
    }
  }
  if (K & 31) {
    const int k = KU + kgrp * 8;
    bf16x8 a = g16_load8(A, arow, as_, k, K, B, arow);
#pragma unroll
    for (int t = 0; t < TILES; ++t) {
      const int ncol0 = (t * 4 + wv) * 16;
      bf16x8 b = g16_load8(W, ncol0 + arow, K, k, K, N, ncol0 + arow);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[t], 0, 0, 0);
    }
  }
#pragma unroll
  for (int t = 0; t < TILES; ++t) {
    const int ncol0 = (t * 4 + wv) * 16;
#pragma unroll
    for (int r = 0; r < 4; ++r) lds[(kgrp * 4 + r) * N + ncol0 + arow] = acc[t][r];
  }
  __syncthreads();
  // rowwise LN stats from the LDS image (4 waves x 4 rows each)
  __shared__ float mr[2][16];
  for (int m = wv * 4; m < wv * 4 + 4; ++m) {
    float s = 0.f, s2 = 0.f;
    for (int j = lane; j < N; j += 64) {
      float v = lds[m * N + j];
      s += v;
      s2 += v * v;
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      s += __shfl_xor(s, off, 64);
      s2 += __shfl_xor(s2, off, 64);
    }
    if (lane == 0) {
      float mean = s / N;
      float var = s2 / N - mean * mean;
      mr[0][m] = mean;
      mr[1][m] = rsqrtf(fmaxf(var, 0.f) + eps);
      if (m < B) {
        mean_out[m] = mean;
        rstd_out[m] = mr[1][m];
      }
    }
  }
  __syncthreads();
  // write G (pre-LN) and Y = silu(LN(G))
  for (int i = threadIdx.x; i < B * N; i += blockDim.x) {
    const int m = i / N, j = i - m * N;
    const float v = lds[m * N + j];
    G[(long)m * gs + j] = (__hip_bfloat16)v;
    float z = (v - mr[0][m]) * mr[1][m] * __bfloat162float(lnw[j]) + __bfloat162float(lnb[j]);
    Y[(long)m * ys + j] = (__hip_bfloat16)(z / (1.f + expf(-z)));
  }
}

void g16_ln_silu(const torch::Tensor& A, const torch::Tensor& W, const torch::Tensor& lnw,
                 const torch::Tensor& lnb, torch::Tensor G, torch::Tensor Y, torch::Tensor mean,
                 torch::Tensor rstd, double eps) {
  TORCH_CHECK(A.dim() == 2 && A.stride(1) == 1 && W.is_contiguous() && A.scalar_type() == at::kBFloat16);
  int B = (int)A.size(0), K = (int)A.size(1), N = (int)W.size(0);
  TORCH_CHECK(B <= 16 && N % 64 == 0 && N <= 512 && G.stride(1) == 1 && Y.stride(1) == 1);
  auto stream = at::cuda::getCurrentCUDAStream();
#define SHEEP_G16LN_CASE(TV)                                                                              \
  case TV:                                                                                                \
    hipLaunchKernelGGL(g16_ln_silu_kernel<TV>, dim3(1), dim3(256), 16 * N * sizeof(float),                \
                       stream.stream(), (const __hip_bfloat16*)A.data_ptr(), A.stride(0),                 \
                       (const __hip_bfloat16*)W.data_ptr(), (const __hip_bfloat16*)lnw.data_ptr(),        \
                       (const __hip_bfloat16*)lnb.data_ptr(), (__hip_bfloat16*)G.data_ptr(), G.stride(0), \
                       (__hip_bfloat16*)Y.data_ptr(), Y.stride(0), mean.data_ptr<float>(),                \
                       rstd.data_ptr<float>(), B, N, K, (float)eps);                                      \
    break;
  switch (N / 64) {
    SHEEP_G16LN_CASE(1)
    SHEEP_G16LN_CASE(2)
    SHEEP_G16LN_CASE(4)
    SHEEP_G16LN_CASE(8)
    default:
      TORCH_CHECK(false, "g16_ln_silu: unsupported N");
  }
#undef SHEEP_G16LN_CASE
}

// single-workgroup GEMM + bias + unimix categorical-ST epilogue (the
// posterior head): raw = A @ W^T + b, then per-group softmax / log-prob /
// gumbel one-hot straight into the stacked m/z/s buffers.  N <= 1024.
template <int TILES>
__global__ void __launch_bounds__(256) g16_cat_st_kernel(const __hip_bfloat16* __restrict__ A, long as_,
                                                         const __hip_bfloat16* __restrict__ W,
                                                         const __hip_bfloat16* __restrict__ bias,
                                                         const float* __restrict__ urand,
                                                         float* __restrict__ m_out,
                                                         __hip_bfloat16* __restrict__ z_out,
                                                         float* __restrict__ s_out, int B, int N, int K, int KD,
                                                         float unimix) {
  extern __shared__ __attribute__((aligned(16))) float lds[];  // [16][N]
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int wv = threadIdx.x >> 6;
  f32x4 acc[TILES];
#pragma unroll
  for (int t = 0; t < TILES; ++t) acc[t] = (f32x4){0.f, 0.f, 0.f, 0.f};
  const int KU = K & ~31;
  const int arow_c = arow < B ? arow : B - 1;
#pragma unroll 4
  for (int k0 = 0; k0 < KU; k0 += 32) {
    const int k = k0 + kgrp * 8;
    bf16x8 a = g16_load8v(A, arow_c, as_, k);
#pragma unroll
    for (int t = 0; t < TILES; ++t) {
      const int ncol0 = (t * 4 + wv) * 16;
      bf16x8 b = g16_load8v(W, ncol0 + arow, K, k);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[t], 0, 0, 0);
    }
  }
  if (K & 31) {
    const int k = KU + kgrp * 8;
    bf16x8 a = g16_load8(A, arow, as_, k, K, B, arow);
#pragma unroll
    for (int t = 0; t < TILES; ++t) {
      const int ncol0 = (t * 4 + wv) * 16;
      bf16x8 b = g16_load8(W, ncol0 + arow, K, k, K, N, ncol0 + arow);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[t], 0, 0, 0);
    }
  }
#pragma unroll
  for (int t = 0; t < TILES; ++t) {
    const int ncol0 = (t * 4 + wv) * 16;
#pragma unroll
    for (int r = 0; r < 4; ++r)
      lds[(kgrp * 4 + r) * N + ncol0 + arow] = acc[t][r] + __bfloat162float(bias[ncol0 + arow]);
  }
  __syncthreads();
  const int groups = N / KD;
  for (int task = threadIdx.x; task < 16 * groups; task += blockDim.x) {
    const int m = task / groups;
    if (m >= B) continue;
    const int g = task - m * groups;
    const float* row = lds + m * N + g * KD;
    float lmax = -1e30f;
    for (int j = 0; j < KD; ++j) lmax = fmaxf(lmax, row[j]);
    float lsum = 0.f;
    for (int j = 0; j < KD; ++j) lsum += expf(row[j] - lmax);
    const float inv = 1.f / lsum;
    const float* ur = urand + ((long)m * groups + g) * KD;
    float* mr_ = m_out + (long)m * N + g * KD;
    float* sr = s_out + (long)m * N + g * KD;
    __hip_bfloat16* zr = z_out + (long)m * N + g * KD;
    float best = -1e30f;
    int best_j = 0;
    for (int j = 0; j < KD; ++j) {
      float sv = expf(row[j] - lmax) * inv;
      float pv = (1.f - unimix) * sv + unimix / KD;
      float mv = logf(pv);
      sr[j] = sv;
      mr_[j] = mv;
      float tt = fmaxf(-logf(fmaxf(ur[j], 1e-20f)), 1e-20f);
      float score = mv - logf(tt);
      if (score > best) {
        best = score;
        best_j = j;
      }
    }
    for (int j = 0; j < KD; ++j) zr[j] = (__hip_bfloat16)(j == best_j ? 1.f : 0.f);
  }
}

void g16_cat_st(const torch::Tensor& A, const torch::Tensor& W, const torch::Tensor& bias,
                const torch::Tensor& urand, torch::Tensor m, torch::Tensor z, torch::Tensor s, long KD,
                double unimix) {
  TORCH_CHECK(A.dim() == 2 && A.stride(1) == 1 && W.is_contiguous() && A.scalar_type() == at::kBFloat16);
  int B = (int)A.size(0), K = (int)A.size(1), N = (int)W.size(0);
  TORCH_CHECK(B <= 16 && N % 64 == 0 && N <= 1024 && m.is_contiguous() && z.is_contiguous() &&
              s.is_contiguous() && N % KD == 0);
  auto stream = at::cuda::getCurrentCUDAStream();
#define SHEEP_G16CS_CASE(TV)                                                                             \
  case TV:                                                                                               \
    hipLaunchKernelGGL(g16_cat_st_kernel<TV>, dim3(1), dim3(256), 16 * N * sizeof(float),                \
                       stream.stream(), (const __hip_bfloat16*)A.data_ptr(), A.stride(0),                \
                       (const __hip_bfloat16*)W.data_ptr(), (const __hip_bfloat16*)bias.data_ptr(),      \
                       urand.data_ptr<float>(), m.data_ptr<float>(), (__hip_bfloat16*)z.data_ptr(),      \
                       s.data_ptr<float>(), B, N, K, (int)KD, (float)unimix);                            \
    break;
  switch (N / 64) {
    SHEEP_G16CS_CASE(1)
    SHEEP_G16CS_CASE(2)
    SHEEP_G16CS_CASE(4)
    SHEEP_G16CS_CASE(8)
    SHEEP_G16CS_CASE(16)
    default:
      TORCH_CHECK(false, "g16_cat_st: unsupported N");
  }
#undef SHEEP_G16CS_CASE
}

// Micro-kernel used by the layout unit test: Y[16,N] = X[16,K] @ W[N,K]^T.
__global__ void pk_gemm16_test_kernel(const __hip_bfloat16* __restrict__ X, const __hip_bfloat16* __restrict__ W,
                                      float* __restrict__ Y, int N, int K) {
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int ncol0 = (blockIdx.x * (int)(blockDim.x >> 6) + wv) * 16;
  if (ncol0 >= N) return;
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  for (int k0 = 0; k0 < K; k0 += 32) {
    bf16x8 a = *(const bf16x8*)(X + (long)arow * K + k0 + kgrp * 8);
    bf16x8 b = *(const bf16x8*)(W + (long)(ncol0 + arow) * K + k0 + kgrp * 8);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) Y[(long)(kgrp * 4 + r) * N + ncol0 + (lane & 15)] = acc[r];
}

torch::Tensor pk_gemm16_test(const torch::Tensor& x, const torch::Tensor& w) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous());
  TORCH_CHECK(x.size(0) == 16 && x.size(1) == w.size(1) && w.size(1) % 32 == 0 && w.size(0) % 64 == 0);
  int K = (int)x.size(1), N = (int)w.size(0);
  auto y = torch::empty({16, N}, x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(pk_gemm16_test_kernel, dim3(N / 64), dim3(256), 0, stream.stream(),
                     (const __hip_bfloat16*)x.data_ptr(), (const __hip_bfloat16*)w.data_ptr(),
                     y.data_ptr<float>(), N, K);
  return y;
}

// grid-wide barrier for persistent kernels: all workgroups must be resident
// (grid <= #CUs).  counter/gen live in a small global workspace.
// Cross-XCD visibility: every wave drains its stores (vmcnt), the arriving
// thread issues an agent-scope RELEASE fence before the ticket and the
// leaving threads an agent-scope ACQUIRE fence after the spin (per
// cdna_hip_programming.md guideline 16 — cheaper than __threadfence()'s
// full writeback on both sides).
__device__ __forceinline__ void pk_grid_barrier(int* counter, volatile int* gen, int nwg) {
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    const int g = __hip_atomic_load((int*)gen, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
    if (__hip_atomic_fetch_add(counter, 1, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT) == nwg - 1) {
      __hip_atomic_store(counter, 0, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      __hip_atomic_fetch_add((int*)gen, 1, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    } else {
      while (__hip_atomic_load((int*)gen, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT) == g) {
      }
    }
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  }
  __syncthreads();
}

// barrier smoke test: NWG workgroups increment a per-round slot ITERS times;
// result[i] must equal nwg for every round.
__global__ void pk_barrier_test_kernel(int* ws, int* out, int nwg, int iters) {
  for (int i = 0; i < iters; ++i) {
    if (threadIdx.x == 0) atomicAdd(&out[i], 1);
    pk_grid_barrier(ws, (volatile int*)(ws + 1), nwg);
    // after the barrier every WG must see the full count of round i
    if (threadIdx.x == 0 && out[i] != nwg) out[iters] = 1 + i;  // flag failure
    pk_grid_barrier(ws, (volatile int*)(ws + 1), nwg);
  }
}

torch::Tensor pk_barrier_test(long nwg, long iters) {
  auto ws = torch::zeros({2}, torch::TensorOptions().dtype(at::kInt).device(at::kCUDA));
  auto out = torch::zeros({iters + 1}, torch::TensorOptions().dtype(at::kInt).device(at::kCUDA));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(pk_barrier_test_kernel, dim3((int)nwg), dim3(256), 0, stream.stream(),
                     ws.data_ptr<int>(), out.data_ptr<int>(), (int)nwg, (int)iters);
  return out;
}

// ---------------------------------------------------------------------------
// Persistent fused RSSM scan FORWARD: the whole T-step recurrence in ONE
// kernel launch.  The per-step launch sequence (3 masked-lerp, 3-4 GEMMs,
// 2 LN, GRU gates, categorical-ST ~ 10 kernels x ~5 us latency each) is
// replaced by grid-resident workgroups that cycle through five phases per
// step, separated by a device-wide barrier (~2.7 us):
//   PA: G1 = X @ W1^T            (X assembled inline from the episode resets)
//   PB: Y  = [h', silu(LN(G1))] @ W2^T
//   PB2: GRU gate math -> h_t    (elementwise, needs the full-row LN sums)
//   PC: G3 = [h_t, embed] @ W3^T
//   PD: RAW = silu(LN(G3)) @ W4^T + b4 -> unimix softmax + gumbel one-hot
// GEMMs run on v_mfma_f32_16x16x32_bf16 with M=16 row tiles (the DV3 batch),
// one 16-col tile per wave, 64-col tile per workgroup; weights stream from
// L2.  All intermediates are written to the SAME stacked [T, B, *] buffers
// the multi-kernel scan uses, so the hand-written backward is unchanged.
// bf16-only, B <= 16; the Python wrapper falls back to the multi-kernel path
// otherwise.

struct PkParams {
  const __hip_bfloat16 *actions, *f_all, *ih, *iz;
  const __hip_bfloat16 *w1, *lnw1, *lnb1, *w2, *lnwg, *lnbg, *w3, *lnw3, *lnb3, *w4, *b4;
  const float* urand;
  __hip_bfloat16 *x_s, *g1_s, *hu_s, *y_s, *r_s, *g3_s, *p_s, *h_seq, *z_seq;
  float *mr1, *mrg, *mr3, *m_seq, *s_s;  // mr*: [2,T,B]
  float* ws;   // [2 slots][3 lns][2 sums][16 rows]
  int* ibar;   // {counter, gen}
  int T, B, A, H, SK, D, P, E, K1p, S, KD, nwg;
  float unimix, eps;
};

__device__ __forceinline__ float pk_b2f(__hip_bfloat16 v) { return __bfloat162float(v); }
__device__ __forceinline__ float pk_h2f(__bf16 v) { return (float)v; }

// guarded, lerp-transformed load of X[m][k0..k0+7] (the G1 GEMM A operand):
// X = [ (1-f)z_prev + f*iz , (1-f)a ] padded with zeros to K1p
__device__ bf16x8 pk_load_x(const PkParams& P, int t, int m, int k0) {
  bf16x8 out;
  if (m >= P.B) {
#pragma unroll
    for (int e = 0; e < 8; ++e) out[e] = (__bf16)0.f;
    return out;
  }
  const float f = pk_b2f(P.f_all[t * P.B + m]);
  if (k0 + 8 <= P.SK) {
    bf16x8 zi = *(const bf16x8*)(P.iz + (long)m * P.SK + k0);
    if (t == 0) {
#pragma unroll
      for (int e = 0; e < 8; ++e) out[e] = (__bf16)(f * pk_h2f(zi[e]));
    } else {
      bf16x8 zp = *(const bf16x8*)(P.z_seq + ((long)(t - 1) * P.B + m) * P.SK + k0);
#pragma unroll
      for (int e = 0; e < 8; ++e) out[e] = (__bf16)((1.f - f) * pk_h2f(zp[e]) + f * pk_h2f(zi[e]));
    }
    return out;
  }
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    const int k = k0 + e;
    float v = 0.f;
    if (k < P.SK) {
      float zp = t > 0 ? pk_b2f(P.z_seq[((long)(t - 1) * P.B + m) * P.SK + k]) : 0.f;
      v = (1.f - f) * zp + f * pk_b2f(P.iz[(long)m * P.SK + k]);
    } else if (k < P.SK + P.A) {
      v = (1.f - f) * pk_b2f(P.actions[((long)t * P.B + m) * P.A + (k - P.SK)]);
    }
    out[e] = (__bf16)v;
  }
  return out;
}

// silu(LN(x)) for one element given the row mean / rstd
__device__ __forceinline__ float pk_silu_ln(float x, float mean, float rstd, float w, float b) {
  float z = (x - mean) * rstd * w + b;
  return z / (1.f + expf(-z));
}

__global__ void __launch_bounds__(256) pk_scan_fwd_kernel(PkParams P) {
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int wg = blockIdx.x;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int tb = P.T * P.B;
  extern __shared__ __attribute__((aligned(16))) float lds[];  // [16][64] raw tile (PD)

  for (int t = 0; t < P.T; ++t) {
    const int slot = t & 1;
    float* acc_g1 = P.ws + ((slot * 3 + 0) * 2) * 16;
    float* acc_y = P.ws + ((slot * 3 + 1) * 2) * 16;
    float* acc_g3 = P.ws + ((slot * 3 + 2) * 2) * 16;

    // ---------------- PA: G1 = X @ W1^T ----------------
    if (wg < P.D / 64) {
      const int ncol0 = wg * 64 + wv * 16;
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      for (int k0 = 0; k0 < P.K1p; k0 += 32) {
        bf16x8 a = pk_load_x(P, t, arow, k0 + kgrp * 8);
        bf16x8 b;
        {
          const int wrow = ncol0 + arow;
          const int k = k0 + kgrp * 8;
          if (k + 8 <= P.SK + P.A) {
            b = *(const bf16x8*)(P.w1 + (long)wrow * (P.SK + P.A) + k);
          } else {
#pragma unroll
            for (int e = 0; e < 8; ++e)
              b[e] = (k + e < P.SK + P.A) ? (__bf16)pk_b2f(P.w1[(long)wrow * (P.SK + P.A) + k + e]) : (__bf16)0.f;
          }
        }
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
      }
      // store g1 tile + row partial sums
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = kgrp * 4 + r;
        if (m < P.B) P.g1_s[((long)t * P.B + m) * P.D + ncol0 + arow] = (__hip_bfloat16)acc[r];
        float v = acc[r];
        float v2 = v * v;
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) {
          v += __shfl_xor(v, off, 64);
          v2 += __shfl_xor(v2, off, 64);
        }
        if (arow == 0 && m < P.B) {
          atomicAdd(&acc_g1[m], v);
          atomicAdd(&acc_g1[16 + m], v2);
        }
      }
    } else if (wg == P.D / 64) {
      // assemble h' into hu_s[t][:, :H] (needed by PB and saved for backward)
      for (int i = threadIdx.x; i < P.B * P.H; i += blockDim.x) {
        const int m = i / P.H, j = i - m * P.H;
        const float f = pk_b2f(P.f_all[t * P.B + m]);
        float hp = t > 0 ? pk_b2f(P.h_seq[((long)(t - 1) * P.B + m) * P.H + j]) : 0.f;
        float v = (1.f - f) * hp + f * pk_b2f(P.ih[(long)m * P.H + j]);
        P.hu_s[((long)t * P.B + m) * (P.H + P.D) + j] = (__hip_bfloat16)v;
      }
    } else if (wg == P.D / 64 + 1) {
      // store x_s[t] for the backward pass
      for (int i = threadIdx.x; i < P.B * (P.SK + P.A); i += blockDim.x) {
        const int m = i / (P.SK + P.A), k = i - m * (P.SK + P.A);
        const float f = pk_b2f(P.f_all[t * P.B + m]);
        float v;
        if (k < P.SK) {
          float zp = t > 0 ? pk_b2f(P.z_seq[((long)(t - 1) * P.B + m) * P.SK + k]) : 0.f;
          v = (1.f - f) * zp + f * pk_b2f(P.iz[(long)m * P.SK + k]);
        } else {
          v = (1.f - f) * pk_b2f(P.actions[((long)t * P.B + m) * P.A + (k - P.SK)]);
        }
        P.x_s[(long)t * P.B * (P.SK + P.A) + i] = (__hip_bfloat16)v;
      }
    }
    pk_grid_barrier(P.ibar, (volatile int*)(P.ibar + 1), P.nwg);

    // ---------------- PB: Y = HU @ W2^T ----------------
    {
      const float m_g1 = acc_g1[arow] / P.D;
      const float var_g1 = acc_g1[16 + arow] / P.D - m_g1 * m_g1;
      const float rs_g1 = rsqrtf(fmaxf(var_g1, 0.f) + P.eps);
      if (wg == 0 && wv == 0 && lane < P.B) {
        P.mr1[(long)t * P.B + lane] = acc_g1[lane] / P.D;
        float mm = acc_g1[lane] / P.D;
        float vv = acc_g1[16 + lane] / P.D - mm * mm;
        P.mr1[(long)tb + t * P.B + lane] = rsqrtf(fmaxf(vv, 0.f) + P.eps);
      }
      if (wg < (3 * P.H) / 64) {
        const int ncol0 = wg * 64 + wv * 16;
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        const int KK = P.H + P.D;
        for (int k0 = 0; k0 < KK; k0 += 32) {
          const int k = k0 + kgrp * 8;
          bf16x8 a;
          if (arow >= P.B) {
#pragma unroll
            for (int e = 0; e < 8; ++e) a[e] = (__bf16)0.f;
          } else if (k + 8 <= P.H) {
            a = *(const bf16x8*)(P.hu_s + ((long)t * P.B + arow) * (long)KK + k);
          } else {
            // u region: silu(LN(g1)) computed inline (k >= H; H%32==0 so no straddle)
            bf16x8 g = *(const bf16x8*)(P.g1_s + ((long)t * P.B + arow) * P.D + (k - P.H));
#pragma unroll
            for (int e = 0; e < 8; ++e) {
              const int j = k - P.H + e;
              a[e] = (__bf16)pk_silu_ln(pk_h2f(g[e]), m_g1, rs_g1, pk_b2f(P.lnw1[j]), pk_b2f(P.lnb1[j]));
            }
          }
          bf16x8 b = *(const bf16x8*)(P.w2 + (long)(ncol0 + arow) * KK + k);
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int m = kgrp * 4 + r;
          if (m < P.B) P.y_s[((long)t * P.B + m) * (3 * P.H) + ncol0 + arow] = (__hip_bfloat16)acc[r];
          float v = acc[r], v2 = v * v;
#pragma unroll
          for (int off = 1; off < 16; off <<= 1) {
            v += __shfl_xor(v, off, 64);
            v2 += __shfl_xor(v2, off, 64);
          }
          if (arow == 0 && m < P.B) {
            atomicAdd(&acc_y[m], v);
            atomicAdd(&acc_y[16 + m], v2);
          }
        }
      }
      // save u into hu_s[t][:, H:]
      if (wg < P.D / 64) {
        for (int i = threadIdx.x; i < P.B * 64; i += blockDim.x) {
          const int m = i >> 6, j0 = wg * 64 + (i & 63);
          const float mm = acc_g1[m] / P.D;
          const float vv = acc_g1[16 + m] / P.D - mm * mm;
          const float rs = rsqrtf(fmaxf(vv, 0.f) + P.eps);
          float g = pk_b2f(P.g1_s[((long)t * P.B + m) * P.D + j0]);
          P.hu_s[((long)t * P.B + m) * (P.H + P.D) + P.H + j0] =
              (__hip_bfloat16)pk_silu_ln(g, mm, rs, pk_b2f(P.lnw1[j0]), pk_b2f(P.lnb1[j0]));
        }
      }
    }
    pk_grid_barrier(P.ibar, (volatile int*)(P.ibar + 1), P.nwg);

    // ---------------- PB2: GRU gates -> h_t ----------------
    if (wg < P.H / 64) {
      for (int i = threadIdx.x; i < P.B * 64; i += blockDim.x) {
        const int m = i >> 6, j = wg * 64 + (i & 63);
        const int D3 = 3 * P.H;
        const float mean = acc_y[m] / D3;
        const float var = acc_y[16 + m] / D3 - mean * mean;
        const float rs = rsqrtf(fmaxf(var, 0.f) + P.eps);
        const __hip_bfloat16* yr = P.y_s + ((long)t * P.B + m) * D3;
        float zr = (pk_b2f(yr[j]) - mean) * rs * pk_b2f(P.lnwg[j]) + pk_b2f(P.lnbg[j]);
        float zc = (pk_b2f(yr[P.H + j]) - mean) * rs * pk_b2f(P.lnwg[P.H + j]) + pk_b2f(P.lnbg[P.H + j]);
        float zu = (pk_b2f(yr[2 * P.H + j]) - mean) * rs * pk_b2f(P.lnwg[2 * P.H + j]) + pk_b2f(P.lnbg[2 * P.H + j]);
        float r = 1.f / (1.f + expf(-zr));
        float c = tanhf(r * zc);
        float u = 1.f / (1.f + expf(-(zu - 1.f)));
        float hp = pk_b2f(P.hu_s[((long)t * P.B + m) * (P.H + P.D) + j]);
        float hv = u * c + (1.f - u) * hp;
        P.h_seq[((long)t * P.B + m) * P.H + j] = (__hip_bfloat16)hv;
        P.r_s[((long)t * P.B + m) * (P.H + P.E) + j] = (__hip_bfloat16)hv;
      }
      if (wg == 0 && wv == 0 && lane < P.B) {
        const int D3 = 3 * P.H;
        float mm = acc_y[lane] / D3;
        float vv = acc_y[16 + lane] / D3 - mm * mm;
        P.mrg[(long)t * P.B + lane] = mm;
        P.mrg[(long)tb + t * P.B + lane] = rsqrtf(fmaxf(vv, 0.f) + P.eps);
      }
    } else if (wg == P.H / 64) {
      // zero the other slot's accumulators for step t+1
      const int other = (t + 1) & 1;
      for (int i = threadIdx.x; i < 3 * 2 * 16; i += blockDim.x) P.ws[other * 96 + i] = 0.f;
    }
    pk_grid_barrier(P.ibar, (volatile int*)(P.ibar + 1), P.nwg);

    // ---------------- PC: G3 = R @ W3^T ----------------
    if (wg < P.P / 64) {
      const int ncol0 = wg * 64 + wv * 16;
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      const int KK = P.H + P.E;
      for (int k0 = 0; k0 < KK; k0 += 32) {
        const int k = k0 + kgrp * 8;
        bf16x8 a;
        if (arow >= P.B) {
#pragma unroll
          for (int e = 0; e < 8; ++e) a[e] = (__bf16)0.f;
        } else {
          a = *(const bf16x8*)(P.r_s + ((long)t * P.B + arow) * (long)KK + k);
        }
        bf16x8 b = *(const bf16x8*)(P.w3 + (long)(ncol0 + arow) * KK + k);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = kgrp * 4 + r;
        if (m < P.B) P.g3_s[((long)t * P.B + m) * P.P + ncol0 + arow] = (__hip_bfloat16)acc[r];
        float v = acc[r], v2 = v * v;
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) {
          v += __shfl_xor(v, off, 64);
          v2 += __shfl_xor(v2, off, 64);
        }
        if (arow == 0 && m < P.B) {
          atomicAdd(&acc_g3[m], v);
          atomicAdd(&acc_g3[16 + m], v2);
        }
      }
    }
    pk_grid_barrier(P.ibar, (volatile int*)(P.ibar + 1), P.nwg);

    // ---------------- PD: RAW = p @ W4^T + b4 -> categorical ST ----------------
    {
      const float m_g3 = acc_g3[arow] / P.P;
      const float var_g3 = acc_g3[16 + arow] / P.P - m_g3 * m_g3;
      const float rs_g3 = rsqrtf(fmaxf(var_g3, 0.f) + P.eps);
      if (wg == 0 && wv == 0 && lane < P.B) {
        float mm = acc_g3[lane] / P.P;
        float vv = acc_g3[16 + lane] / P.P - mm * mm;
        P.mr3[(long)t * P.B + lane] = mm;
        P.mr3[(long)tb + t * P.B + lane] = rsqrtf(fmaxf(vv, 0.f) + P.eps);
      }
      if (wg < P.P / 64) {
        // save p for the backward pass
        for (int i = threadIdx.x; i < P.B * 64; i += blockDim.x) {
          const int m = i >> 6, j0 = wg * 64 + (i & 63);
          const float mm = acc_g3[m] / P.P;
          const float vv = acc_g3[16 + m] / P.P - mm * mm;
          const float rs = rsqrtf(fmaxf(vv, 0.f) + P.eps);
          float g = pk_b2f(P.g3_s[((long)t * P.B + m) * P.P + j0]);
          P.p_s[((long)t * P.B + m) * P.P + j0] =
              (__hip_bfloat16)pk_silu_ln(g, mm, rs, pk_b2f(P.lnw3[j0]), pk_b2f(P.lnb3[j0]));
        }
      }
      if (wg < P.SK / 64) {
        const int ncol0 = wg * 64 + wv * 16;
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        for (int k0 = 0; k0 < P.P; k0 += 32) {
          const int k = k0 + kgrp * 8;
          bf16x8 a;
          if (arow >= P.B) {
#pragma unroll
            for (int e = 0; e < 8; ++e) a[e] = (__bf16)0.f;
          } else {
            bf16x8 g = *(const bf16x8*)(P.g3_s + ((long)t * P.B + arow) * P.P + k);
#pragma unroll
            for (int e = 0; e < 8; ++e)
              a[e] = (__bf16)pk_silu_ln(pk_h2f(g[e]), m_g3, rs_g3, pk_b2f(P.lnw3[k + e]), pk_b2f(P.lnb3[k + e]));
          }
          bf16x8 b = *(const bf16x8*)(P.w4 + (long)(ncol0 + arow) * P.P + k);
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
        }
        // raw tile (+bias) to LDS, then per-(row, group) categorical
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int m = kgrp * 4 + r;
          lds[m * 64 + wv * 16 + arow] = acc[r] + pk_b2f(P.b4[ncol0 + arow]);
        }
        __syncthreads();
        const int groups_per_wg = 64 / P.KD;  // KD divides 64
        const int tasks = 16 * groups_per_wg;
        for (int task = threadIdx.x; task < tasks; task += blockDim.x) {
          const int m = task / groups_per_wg;
          if (m >= P.B) continue;
          const int gloc = task - m * groups_per_wg;
          const int col0 = gloc * P.KD;                 // within the WG tile
          const int gcol0 = wg * 64 + col0;             // absolute column
          const int sgrp = gcol0 / P.KD;                // stochastic group index
          float lmax = -1e30f;
          for (int j = 0; j < P.KD; ++j) lmax = fmaxf(lmax, lds[m * 64 + col0 + j]);
          float lsum = 0.f;
          for (int j = 0; j < P.KD; ++j) lsum += expf(lds[m * 64 + col0 + j] - lmax);
          const float inv = 1.f / lsum;
          float best = -1e30f;
          int best_j = 0;
          const float* ur = P.urand + (((long)t * P.B + m) * P.S + sgrp) * P.KD;
          float* mr = P.m_seq + ((long)t * P.B + m) * P.SK + gcol0;
          float* sr = P.s_s + ((long)t * P.B + m) * P.SK + gcol0;
          __hip_bfloat16* zr = P.z_seq + ((long)t * P.B + m) * P.SK + gcol0;
          for (int j = 0; j < P.KD; ++j) {
            float sv = expf(lds[m * 64 + col0 + j] - lmax) * inv;
            float pv = (1.f - P.unimix) * sv + P.unimix / P.KD;
            float mv = logf(pv);
            sr[j] = sv;
            mr[j] = mv;
            float tt = fmaxf(-logf(fmaxf(ur[j], 1e-20f)), 1e-20f);
            float score = mv - logf(tt);
            if (score > best) {
              best = score;
              best_j = j;
            }
          }
          for (int j = 0; j < P.KD; ++j) zr[j] = (__hip_bfloat16)(j == best_j ? 1.f : 0.f);
        }
        __syncthreads();
      }
    }
    pk_grid_barrier(P.ibar, (volatile int*)(P.ibar + 1), P.nwg);
  }
}

void pk_scan_fwd(const torch::Tensor& actions, const torch::Tensor& f_all, const torch::Tensor& ih,
                 const torch::Tensor& iz, const torch::Tensor& w1, const torch::Tensor& lnw1,
                 const torch::Tensor& lnb1, const torch::Tensor& w2, const torch::Tensor& lnwg,
                 const torch::Tensor& lnbg, const torch::Tensor& w3, const torch::Tensor& lnw3,
                 const torch::Tensor& lnb3, const torch::Tensor& w4, const torch::Tensor& b4,
                 const torch::Tensor& urand, torch::Tensor x_s, torch::Tensor g1_s, torch::Tensor hu_s,
                 torch::Tensor y_s, torch::Tensor r_s, torch::Tensor g3_s, torch::Tensor p_s,
                 torch::Tensor h_seq, torch::Tensor z_seq, torch::Tensor mr1, torch::Tensor mrg,
                 torch::Tensor mr3, torch::Tensor m_seq, torch::Tensor s_s, torch::Tensor ws,
                 torch::Tensor ibar, double unimix, double eps) {
  PkParams P;
  P.T = (int)h_seq.size(0);
  P.B = (int)h_seq.size(1);
  P.H = (int)h_seq.size(2);
  P.SK = (int)z_seq.size(2);
  P.A = (int)actions.size(2);
  P.D = (int)g1_s.size(2);
  P.P = (int)g3_s.size(2);
  P.E = (int)r_s.size(2) - P.H;
  P.K1p = ((P.SK + P.A + 31) / 32) * 32;
  P.KD = (int)s_s.size(3);
  P.S = P.SK / P.KD;
  P.unimix = (float)unimix;
  P.eps = (float)eps;
  TORCH_CHECK(P.B <= 16 && w1.scalar_type() == at::kBFloat16, "pk_scan_fwd: bf16, B<=16 only");
  TORCH_CHECK(P.D % 64 == 0 && P.H % 64 == 0 && P.P % 64 == 0 && P.SK % 64 == 0, "pk dims");
  TORCH_CHECK((P.H + P.D) % 32 == 0 && (P.H + P.E) % 32 == 0 && P.P % 32 == 0 && P.SK % 32 == 0, "pk K dims");
  TORCH_CHECK(P.KD <= 64 && 64 % P.KD == 0, "pk discrete size");
  int nwg = std::max({3 * P.H / 64, P.D / 64 + 2, P.H / 64 + 1, P.P / 64, P.SK / 64});
  TORCH_CHECK(nwg <= 128, "pk grid too large for co-residency");
  P.nwg = nwg;
  P.actions = (const __hip_bfloat16*)actions.data_ptr();
  P.f_all = (const __hip_bfloat16*)f_all.data_ptr();
  P.ih = (const __hip_bfloat16*)ih.data_ptr();
  P.iz = (const __hip_bfloat16*)iz.data_ptr();
  P.w1 = (const __hip_bfloat16*)w1.data_ptr();
  P.lnw1 = (const __hip_bfloat16*)lnw1.data_ptr();
  P.lnb1 = (const __hip_bfloat16*)lnb1.data_ptr();
  P.w2 = (const __hip_bfloat16*)w2.data_ptr();
  P.lnwg = (const __hip_bfloat16*)lnwg.data_ptr();
  P.lnbg = (const __hip_bfloat16*)lnbg.data_ptr();
  P.w3 = (const __hip_bfloat16*)w3.data_ptr();
  P.lnw3 = (const __hip_bfloat16*)lnw3.data_ptr();
  P.lnb3 = (const __hip_bfloat16*)lnb3.data_ptr();
  P.w4 = (const __hip_bfloat16*)w4.data_ptr();
  P.b4 = (const __hip_bfloat16*)b4.data_ptr();
  P.urand = urand.data_ptr<float>();
  P.x_s = (__hip_bfloat16*)x_s.data_ptr();
  P.g1_s = (__hip_bfloat16*)g1_s.data_ptr();
  P.hu_s = (__hip_bfloat16*)hu_s.data_ptr();
  P.y_s = (__hip_bfloat16*)y_s.data_ptr();
  P.r_s = (__hip_bfloat16*)r_s.data_ptr();
  P.g3_s = (__hip_bfloat16*)g3_s.data_ptr();
  P.p_s = (__hip_bfloat16*)p_s.data_ptr();
  P.h_seq = (__hip_bfloat16*)h_seq.data_ptr();
  P.z_seq = (__hip_bfloat16*)z_seq.data_ptr();
  P.mr1 = mr1.data_ptr<float>();
  P.mrg = mrg.data_ptr<float>();
  P.mr3 = mr3.data_ptr<float>();
  P.m_seq = m_seq.data_ptr<float>();
  P.s_s = s_s.data_ptr<float>();
  P.ws = ws.data_ptr<float>();
  P.ibar = ibar.data_ptr<int>();
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(pk_scan_fwd_kernel, dim3(nwg), dim3(256), 16 * 64 * sizeof(float), stream.stream(), P);
}

// ---------------------------------------------------------------------------
// scan2: multi-workgroup fused phase kernels for the RSSM scan (round 2).
//
// The round-1 scan runs ~9 launches per step (4 hipblaslt GEMMs at the
// ~5.8 us M=16 latency floor + LN/gate/sample epilogue kernels).  These
// kernels fuse each phase into ONE launch: the GEMM runs on the g16 MFMA
// skeleton with the A-operand staged in LDS, and the epilogue (LayerNorm,
// Hafner gates, categorical-ST, episode resets) runs in the same launch.
// LayerNorm statistics that span the N-split workgroups are exchanged with
// the in-launch last-arriver pattern (atomicAdd partials -> ticket ->
// relaxed poll -> ONE agent acquire -> plain reads) per the CDNA4 guide;
// backward kernels avoid any sync by recomputing the small row-local
// gradients per workgroup (the "recompute beats cross-WG traffic" rule for
// B<=16 rows).  Grids are <= H/16 <= 256 workgroups, always co-resident,
// so the bounded spin cannot deadlock.
// ---------------------------------------------------------------------------

__device__ __forceinline__ void sc2_flush_and_arrive(int* ticket) {
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) __hip_atomic_fetch_add(ticket, 1, __ATOMIC_ACQ_REL, __HIP_MEMORY_SCOPE_AGENT);
}

__device__ __forceinline__ void sc2_wait(int* ticket, int nwg) {
  if (threadIdx.x == 0) {
    int spins = 0;
    // bounded spin: never hang the box (a lost ticket produces wrong numbers
    // that the numerics tests catch, not a dead GPU)
    while (__hip_atomic_load(ticket, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT) < nwg &&
           spins < (1 << 27)) {
      __builtin_amdgcn_s_sleep(2);
      ++spins;
    }
  }
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  __syncthreads();
}

// cooperative vectorized copy of a row-strided bf16 [B, C] into LDS [16][KP]
// (zero-padded rows >= B / cols >= C); requires C % 8 == 0 and ss % 8 == 0.
__device__ __forceinline__ void sc2_stage_vec(const __hip_bfloat16* src, long ss, int B, int C,
                                              __hip_bfloat16* dst, int KP) {
  const int CV = C >> 3, KPV = KP >> 3;
  uint4* d4 = (uint4*)dst;
  for (int i = threadIdx.x; i < 16 * KPV; i += blockDim.x) {
    const int m = i / KPV, c = i - m * KPV;
    uint4 v = {0u, 0u, 0u, 0u};
    if (m < B && c < CV) v = *(const uint4*)(src + (long)m * ss + ((long)c << 3));
    d4[i] = v;
  }
}

// one 16x16 C tile of  A[16, K] @ W[N, K]^T  over k in [kbeg, kend) with A
// staged in LDS ([16][KP], KP = K rounded up to 32, zero-padded) and W
// streamed; wrow clamped so ragged-N grids stay in bounds (garbage cols are
// never written).
__device__ __forceinline__ f32x4 sc2_gemm_tile_range(const __hip_bfloat16* ldsA, int KP,
                                                     const __hip_bfloat16* W, long ws_, int K, int N,
                                                     int ncol0, int kbeg, int kend) {
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int wrow = min(ncol0 + arow, N - 1);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int KU = kbeg + ((kend - kbeg) & ~31);
  const int KU8 = kbeg + ((KU - kbeg) & ~255);
  for (int k0 = kbeg; k0 < KU8; k0 += 256) {
    bf16x8 af[8], bf[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int k = k0 + u * 32 + kgrp * 8;
      af[u] = *(const bf16x8*)(ldsA + arow * KP + k);
      bf[u] = *(const bf16x8*)(W + (long)wrow * ws_ + k);
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[u], bf[u], acc, 0, 0, 0);
  }
#pragma unroll 4
  for (int k0 = KU8; k0 < KU; k0 += 32) {
    const int k = k0 + kgrp * 8;
    bf16x8 a = *(const bf16x8*)(ldsA + arow * KP + k);
    bf16x8 b = *(const bf16x8*)(W + (long)wrow * ws_ + k);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
  if (KU < kend) {
    const int k = KU + kgrp * 8;
    bf16x8 a = *(const bf16x8*)(ldsA + arow * KP + k);  // zero-padded: safe
    bf16x8 b;
    const __hip_bfloat16* p = W + (long)wrow * ws_ + k;
#pragma unroll
    for (int e = 0; e < 8; ++e) b[e] = (k + e < kend) ? (__bf16)__bfloat162float(p[e]) : (__bf16)0.f;
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
  return acc;
}

// same tile over k in [kbeg, kend) with A read DIRECTLY from global memory
// (row stride as_, B rows, zero beyond) — used by the K-split mode where each
// wave touches a distinct K-quarter exactly once (LDS staging buys nothing).
__device__ __forceinline__ f32x4 sc2_gemm_tile_range_g(const __hip_bfloat16* A, long as_, int B,
                                                       const __hip_bfloat16* W, long ws_, int K,
                                                       int N, int ncol0, int kbeg, int kend) {
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int wrow = min(ncol0 + arow, N - 1);
  const int arow_c = arow < B ? arow : B - 1;  // clamp: garbage rows unused
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int KU = kbeg + ((kend - kbeg) & ~31);
  const int KU8 = kbeg + ((KU - kbeg) & ~255);
  for (int k0 = kbeg; k0 < KU8; k0 += 256) {
    bf16x8 af[8], bf[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int k = k0 + u * 32 + kgrp * 8;
      af[u] = *(const bf16x8*)(A + (long)arow_c * as_ + k);
      bf[u] = *(const bf16x8*)(W + (long)wrow * ws_ + k);
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[u], bf[u], acc, 0, 0, 0);
  }
#pragma unroll 4
  for (int k0 = KU8; k0 < KU; k0 += 32) {
    const int k = k0 + kgrp * 8;
    bf16x8 a = *(const bf16x8*)(A + (long)arow_c * as_ + k);
    bf16x8 b = *(const bf16x8*)(W + (long)wrow * ws_ + k);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
  if (KU < kend) {
    const int k = KU + kgrp * 8;
    bf16x8 a, b;
    const __hip_bfloat16* ap = A + (long)arow_c * as_ + k;
    const __hip_bfloat16* p = W + (long)wrow * ws_ + k;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      a[e] = (k + e < kend) ? (__bf16)__bfloat162float(ap[e]) : (__bf16)0.f;
      b[e] = (k + e < kend) ? (__bf16)__bfloat162float(p[e]) : (__bf16)0.f;
    }
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
  // note: C rows m >= B carry clamped-row garbage; every caller guards
  // its writes and statistics with m < B (same contract as the LDS path)
  return acc;
}

__device__ __forceinline__ f32x4 sc2_gemm_tile(const __hip_bfloat16* ldsA, int KP,
                                               const __hip_bfloat16* W, long ws_, int K, int N,
                                               int ncol0) {
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int wrow = min(ncol0 + arow, N - 1);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const int KU = K & ~31;
  const int KU8 = KU & ~255;
  for (int k0 = 0; k0 < KU8; k0 += 256) {
    bf16x8 af[8], bf[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int k = k0 + u * 32 + kgrp * 8;
      af[u] = *(const bf16x8*)(ldsA + arow * KP + k);
      bf[u] = *(const bf16x8*)(W + (long)wrow * ws_ + k);
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[u], bf[u], acc, 0, 0, 0);
  }
#pragma unroll 4
  for (int k0 = KU8; k0 < KU; k0 += 32) {
    const int k = k0 + kgrp * 8;
    bf16x8 a = *(const bf16x8*)(ldsA + arow * KP + k);
    bf16x8 b = *(const bf16x8*)(W + (long)wrow * ws_ + k);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
  if (K & 31) {
    const int k = KU + kgrp * 8;
    bf16x8 a = *(const bf16x8*)(ldsA + arow * KP + k);  // zero-padded: safe
    bf16x8 b;
    const __hip_bfloat16* p = W + (long)wrow * ws_ + k;
#pragma unroll
    for (int e = 0; e < 8; ++e) b[e] = (k + e < K) ? (__bf16)__bfloat162float(p[e]) : (__bf16)0.f;
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
  return acc;
}

// per-wave reduce of a per-(row, col) value over the wave's 16 cols; lanes
// with arow == 0 then hold the per-row result (rows kgrp*4 + r).
__device__ __forceinline__ float sc2_colsum(float v) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// ---------------------------------------------------------------------------
// scan2 forward phase 1/3: C = A @ W^T, then rowwise LN (+SiLU) across the
// whole grid via the ticket exchange.  RESETS=true additionally assembles the
// episode-reset-masked GEMM input x = [z', a'] (phase 1) in LDS and persists
// x / h' for the backward.
// ---------------------------------------------------------------------------
// KSPLIT=false: grid N/64, each of 4 waves owns a separate 16-col tile over
// full K.  KSPLIT=true (long-K shapes): grid N/16, the 4 waves K-split ONE
// 16-col tile and combine through LDS — 4x the weight-stream concurrency.
template <bool RESETS, bool T0, bool KSPLIT>
__global__ void __launch_bounds__(256) scan2_lnsilu_kernel(
    const __hip_bfloat16* __restrict__ a_in, long as_,      // staged A (phase 3) or z_prev (phase 1)
    const __hip_bfloat16* __restrict__ iz,                  // phase 1 only
    const __hip_bfloat16* __restrict__ h_prev,
    const __hip_bfloat16* __restrict__ ih,
    const __hip_bfloat16* __restrict__ act,                 // [B, A]
    const __hip_bfloat16* __restrict__ f,                   // [B]
    const __hip_bfloat16* __restrict__ W, const __hip_bfloat16* __restrict__ lnw,
    const __hip_bfloat16* __restrict__ lnb,
    __hip_bfloat16* __restrict__ x_out, long xs,            // phase 1: x_s[t]
    __hip_bfloat16* __restrict__ hu_out, long hus,          // phase 1: hu_s[t] ([:, :H] = h', [:, H:] = u); phase 3: p out
    __hip_bfloat16* __restrict__ g_out, long gs,            // pre-LN GEMM result
    float* __restrict__ mean_out, float* __restrict__ rstd_out,
    float* __restrict__ ws2, int* __restrict__ ticket,
    int B, int SK, int A, int H, int N, int K, float eps, int hu_off) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // +8-element row pad: a 2-KB-multiple row stride lands every A-fragment
  // lane on the same LDS bank group (multi-way ds_read_b128 conflicts)
  const int KP = (((K + 31) & ~31) + 8);
  __hip_bfloat16* ldsA = (__hip_bfloat16*)smem;                  // [16][KP] (unused in KSPLIT)
  float* lds_sum = (float*)(smem + (KSPLIT ? 4096 : ((16 * KP * 2 + 15) & ~15)));  // [32]
  if (threadIdx.x < 32) lds_sum[threadIdx.x] = 0.f;
  if (RESETS) {
    // x = [(1-f) z_prev + f iz, (1-f) a]; rows >= B and cols >= K zero.
    // Vectorized: SK % 8 == 0 (host-checked), the action/pad tail is scalar.
    const int KPV = KP >> 3, SKV = SK >> 3;
    uint4* d4 = (uint4*)ldsA;
    for (int i = threadIdx.x; i < 16 * KPV; i += blockDim.x) {
      const int m = i / KPV, cv = i - m * KPV;
      LnVec<__hip_bfloat16, 8> v;
      v.u = (uint4){0u, 0u, 0u, 0u};
      if (m < B) {
        const float fb = __bfloat162float(f[m]);
        if (cv < SKV) {
          bf16x8 izv = *(const bf16x8*)(iz + (long)m * SK + ((long)cv << 3));
          if (T0) {
#pragma unroll
            for (int e = 0; e < 8; ++e) v.e[e] = __float2bfloat16(fb * (float)izv[e]);
          } else {
            bf16x8 zpv = *(const bf16x8*)(a_in + (long)m * SK + ((long)cv << 3));
#pragma unroll
            for (int e = 0; e < 8; ++e)
              v.e[e] = __float2bfloat16((1.f - fb) * (float)zpv[e] + fb * (float)izv[e]);
          }
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            const int c = (cv << 3) + e;
            if (c >= SK && c < SK + A)
              v.e[e] = __float2bfloat16((1.f - fb) * __bfloat162float(act[(long)m * A + (c - SK)]));
          }
        }
      }
      d4[i] = v.u;
    }
    __syncthreads();
    // persist x and h' for the backward, striped across the grid (every WG
    // staged the identical LDS image — no single-WG straggler)
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < (long)B * K;
         i += (long)gridDim.x * blockDim.x) {
      const int m = (int)(i / K), c = (int)(i - (long)m * K);
      x_out[(long)m * xs + c] = ldsA[m * KP + c];
    }
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < (long)B * H;
         i += (long)gridDim.x * blockDim.x) {
      const int m = (int)(i / H), j = (int)(i - (long)m * H);
      const float fb = __bfloat162float(f[m]);
      float hp = T0 ? 0.f : __bfloat162float(h_prev[(long)m * H + j]);
      st(hu_out, (long)m * hus + j, (1.f - fb) * hp + fb * __bfloat162float(ih[(long)m * H + j]));
    }
  } else if (!KSPLIT) {
    if ((K & 7) == 0 && (as_ & 7) == 0) {
      sc2_stage_vec(a_in, as_, B, K, ldsA, KP);
    } else {
      for (int i = threadIdx.x; i < 16 * KP; i += blockDim.x) {
        const int m = i / KP, c = i - m * KP;
        ldsA[i] = (m < B && c < K) ? a_in[(long)m * as_ + c] : (__hip_bfloat16)0.f;
      }
    }
    __syncthreads();
  } else {
    __syncthreads();  // lds_sum zero-fill visible before atomics
  }
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int wv = threadIdx.x >> 6;
  const int ncol0 = KSPLIT ? blockIdx.x * 16 : (blockIdx.x * 4 + wv) * 16;
  f32x4 acc;
  if (KSPLIT) {
    const int KQ = (((K + 3) / 4) + 31) & ~31;
    const int kbeg = min(wv * KQ, K);
    const int kend = min(kbeg + KQ, K);
    acc = sc2_gemm_tile_range_g(a_in, as_, B, W, K, K, N, ncol0, kbeg, kend);
    __syncthreads();
    float* comb = (float*)smem;  // [4][16][16] = 4 KB
#pragma unroll
    for (int r = 0; r < 4; ++r) comb[(wv * 16 + kgrp * 4 + r) * 16 + arow] = acc[r];
    __syncthreads();
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = kgrp * 4 + r;
      acc[r] = comb[(m)*16 + arow] + comb[(16 + m) * 16 + arow] + comb[(32 + m) * 16 + arow] +
               comb[(48 + m) * 16 + arow];
    }
  } else {
    acc = sc2_gemm_tile(ldsA, KP, W, K, K, N, ncol0);
  }
  const bool emit = !KSPLIT || wv == 0;  // KSPLIT: wave 0 owns the tile
  // write pre-LN G and accumulate row partials
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = kgrp * 4 + r;
    if (emit && m < B && ncol0 + arow < N) g_out[(long)m * gs + ncol0 + arow] = __float2bfloat16(acc[r]);
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    float v = (ncol0 + arow < N) ? acc[r] : 0.f;
    float cs = sc2_colsum(v);
    float cs2 = sc2_colsum(v * v);
    if (emit && arow == 0) {
      atomicAdd(&lds_sum[kgrp * 4 + r], cs);
      atomicAdd(&lds_sum[16 + kgrp * 4 + r], cs2);
    }
  }
  __syncthreads();
  if (threadIdx.x < 32) atomicAdd(&ws2[threadIdx.x], lds_sum[threadIdx.x]);
  sc2_flush_and_arrive(ticket);
  sc2_wait(ticket, gridDim.x);
  __shared__ float mr_[2][16];
  if (threadIdx.x < 16) {
    float mean = ws2[threadIdx.x] / N;
    float var = ws2[16 + threadIdx.x] / N - mean * mean;
    mr_[0][threadIdx.x] = mean;
    mr_[1][threadIdx.x] = rsqrtf(fmaxf(var, 0.f) + eps);
    if (blockIdx.x == 0 && threadIdx.x < B) {
      mean_out[threadIdx.x] = mean;
      rstd_out[threadIdx.x] = mr_[1][threadIdx.x];
    }
  }
  __syncthreads();
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = kgrp * 4 + r;
    const int j = ncol0 + arow;
    if (emit && m < B && j < N) {
      float z = (acc[r] - mr_[0][m]) * mr_[1][m] * __bfloat162float(lnw[j]) + __bfloat162float(lnb[j]);
      z = z / (1.f + expf(-z));
      st(hu_out, (long)m * hus + hu_off + j, z);
    }
  }
}

// ---------------------------------------------------------------------------
// scan2 forward phase 2: y = hu @ W2^T (three gate stripes per h column),
// LN over 3H via the ticket exchange, then the Hafner GRU gates.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256) scan2_gru_kernel(
    const __hip_bfloat16* __restrict__ hu, long hus,  // [B, H+D] (h' | u)
    const __hip_bfloat16* __restrict__ W2,            // [3H, H+D]
    const __hip_bfloat16* __restrict__ lnw, const __hip_bfloat16* __restrict__ lnb,
    __hip_bfloat16* __restrict__ y_out, long ys2,     // pre-LN y
    __hip_bfloat16* __restrict__ h_out,               // h_seq[t] [B, H] contiguous
    __hip_bfloat16* __restrict__ h_out2, long h2s,    // r_s[t][:, :H]
    float* __restrict__ mean_out, float* __restrict__ rstd_out,
    float* __restrict__ ws2, int* __restrict__ ticket, int B, int H, int D, float eps) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int K = H + D;
  const int KP = ((K + 31) & ~31) + 8;  // +8: LDS bank de-phasing
  __hip_bfloat16* ldsA = (__hip_bfloat16*)smem;
  float* lds_sum = (float*)(smem + ((16 * KP * 2 + 15) & ~15));
  if (threadIdx.x < 32) lds_sum[threadIdx.x] = 0.f;
  if ((K & 7) == 0 && (hus & 7) == 0) {
    sc2_stage_vec(hu, hus, B, K, ldsA, KP);
  } else {
    for (int i = threadIdx.x; i < 16 * KP; i += blockDim.x) {
      const int m = i / KP, c = i - m * KP;
      ldsA[i] = (m < B && c < K) ? hu[(long)m * hus + c] : (__hip_bfloat16)0.f;
    }
  }
  __syncthreads();
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int wv = threadIdx.x >> 6;
  const int c0 = (blockIdx.x * 4 + wv) * 16;  // h-column tile
  f32x4 accg[3];
  float s = 0.f, s2 = 0.f;
#pragma unroll
  for (int g = 0; g < 3; ++g) {
    accg[g] = sc2_gemm_tile(ldsA, KP, W2, K, K, 3 * H, g * H + c0);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = kgrp * 4 + r;
      if (m < B) y_out[(long)m * ys2 + g * H + c0 + arow] = __float2bfloat16(accg[g][r]);
      float cs = sc2_colsum(accg[g][r]);
      float cs2 = sc2_colsum(accg[g][r] * accg[g][r]);
      if (arow == 0) {
        atomicAdd(&lds_sum[kgrp * 4 + r], cs);
        atomicAdd(&lds_sum[16 + kgrp * 4 + r], cs2);
      }
    }
  }
  __syncthreads();
  if (threadIdx.x < 32) atomicAdd(&ws2[threadIdx.x], lds_sum[threadIdx.x]);
  sc2_flush_and_arrive(ticket);
  sc2_wait(ticket, gridDim.x);
  const int DD = 3 * H;
  __shared__ float mr_[2][16];
  if (threadIdx.x < 16) {
    float mean = ws2[threadIdx.x] / DD;
    float var = ws2[16 + threadIdx.x] / DD - mean * mean;
    mr_[0][threadIdx.x] = mean;
    mr_[1][threadIdx.x] = rsqrtf(fmaxf(var, 0.f) + eps);
    if (blockIdx.x == 0 && threadIdx.x < B) {
      mean_out[threadIdx.x] = mean;
      rstd_out[threadIdx.x] = mr_[1][threadIdx.x];
    }
  }
  __syncthreads();
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = kgrp * 4 + r;
    const int j = c0 + arow;
    if (m < B) {
      const float mean = mr_[0][m], rstd = mr_[1][m];
      float zr = (accg[0][r] - mean) * rstd * __bfloat162float(lnw[j]) + __bfloat162float(lnb[j]);
      float zc = (accg[1][r] - mean) * rstd * __bfloat162float(lnw[H + j]) + __bfloat162float(lnb[H + j]);
      float zu = (accg[2][r] - mean) * rstd * __bfloat162float(lnw[2 * H + j]) + __bfloat162float(lnb[2 * H + j]);
      float rg = 1.f / (1.f + expf(-zr));
      float cg = tanhf(rg * zc);
      float ug = 1.f / (1.f + expf(-(zu - 1.f)));
      float hp = __bfloat162float(ldsA[m * KP + j]);  // h' lives in hu[:, :H]
      float hv = ug * cg + (1.f - ug) * hp;
      h_out[(long)m * H + j] = __float2bfloat16(hv);
      h_out2[(long)m * h2s + j] = __float2bfloat16(hv);
    }
  }
}

// ---------------------------------------------------------------------------
// scan2 forward phase 4: raw = p @ W4^T + b4, then the unimix categorical-ST
// head per KD-group (no cross-WG traffic: each WG owns whole groups).
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256) scan2_catst_kernel(
    const __hip_bfloat16* __restrict__ p_in, long ps,  // [B, P]
    const __hip_bfloat16* __restrict__ W4,             // [SK, P]
    const __hip_bfloat16* __restrict__ b4, const float* __restrict__ urand,
    float* __restrict__ m_out, __hip_bfloat16* __restrict__ z_out, float* __restrict__ s_out,
    int B, int P, int SK, int KD, float unimix) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int KP = ((P + 31) & ~31) + 8;  // +8: LDS bank de-phasing
  __hip_bfloat16* ldsA = (__hip_bfloat16*)smem;
  float* raw = (float*)(smem + ((16 * KP * 2 + 15) & ~15));  // [16][64]
  if ((P & 7) == 0 && (ps & 7) == 0) {
    sc2_stage_vec(p_in, ps, B, P, ldsA, KP);
  } else {
    for (int i = threadIdx.x; i < 16 * KP; i += blockDim.x) {
      const int m = i / KP, c = i - m * KP;
      ldsA[i] = (m < B && c < P) ? p_in[(long)m * ps + c] : (__hip_bfloat16)0.f;
    }
  }
  __syncthreads();
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int wv = threadIdx.x >> 6;
  const int ncol0 = (blockIdx.x * 4 + wv) * 16;
  f32x4 acc = sc2_gemm_tile(ldsA, KP, W4, P, P, SK, ncol0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = kgrp * 4 + r;
    raw[m * 64 + (ncol0 & 63) + arow] = acc[r] + __bfloat162float(b4[ncol0 + arow]);
  }
  __syncthreads();
  const int gpw = 64 / KD;  // groups per WG per row
  for (int task = threadIdx.x; task < 16 * gpw; task += blockDim.x) {
    const int m = task / gpw;
    if (m >= B) continue;
    const int gl = task - m * gpw;
    const float* row = raw + m * 64 + gl * KD;
    const long gcol = (long)blockIdx.x * 64 + gl * KD;  // global col of the group
    float lmax = -1e30f;
    for (int j = 0; j < KD; ++j) lmax = fmaxf(lmax, row[j]);
    float lsum = 0.f;
    for (int j = 0; j < KD; ++j) lsum += expf(row[j] - lmax);
    const float inv = 1.f / lsum;
    const float* ur = urand + (long)m * SK + gcol;
    float* mro = m_out + (long)m * SK + gcol;
    float* sro = s_out + (long)m * SK + gcol;
    __hip_bfloat16* zro = z_out + (long)m * SK + gcol;
    float best = -1e30f;
    int best_j = 0;
    for (int j = 0; j < KD; ++j) {
      float sv = expf(row[j] - lmax) * inv;
      float pv = (1.f - unimix) * sv + unimix / KD;
      float mv = logf(pv);
      sro[j] = sv;
      mro[j] = mv;
      float tt = fmaxf(-logf(fmaxf(ur[j], 1e-20f)), 1e-20f);
      float score = mv - logf(tt);
      if (score > best) {
        best = score;
        best_j = j;
      }
    }
    for (int j = 0; j < KD; ++j) zro[j] = (__hip_bfloat16)(j == best_j ? 1.f : 0.f);
  }
}

// ---------------------------------------------------------------------------
// scan2 backward phase 4 (reverse order: runs first): categorical-ST backward
// recomputed per WG into LDS, then the gp = graw @ W4t GEMM tile.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256) scan2_b4_kernel(
    const float* __restrict__ gm,                 // [B, SK] fp32
    const __hip_bfloat16* __restrict__ gon,       // g_z_seq[t]
    const __hip_bfloat16* __restrict__ gon2,      // carry or null
    const float* __restrict__ s_saved,            // [B, SK]
    const __hip_bfloat16* __restrict__ W4t,       // [P, SK] (pre-transposed)
    __hip_bfloat16* __restrict__ graw_out, long gws,  // graw_s[t] (WG 0 writes)
    __hip_bfloat16* __restrict__ gp_out,          // [B, P] scratch
    int B, int SK, int P, int KD, float unimix) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int KP = SK + 8;  // +8: LDS bank de-phasing (SK % 64 == 0)
  __hip_bfloat16* graw = (__hip_bfloat16*)smem;  // [16][SK+8]
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  // wave-per-4-rows coalesced recompute; KD <= 64 groups align to lane
  // halves, so the group softmax-grad dot reduces with in-wave shuffles
  for (int mr2 = 0; mr2 < 4; ++mr2) {
    const int m = wv * 4 + mr2;
    for (int c0 = 0; c0 < SK; c0 += 64) {
      const int c = c0 + lane;
      float t = 0.f, sj = 0.f;
      if (m < B) {
        const long idx = (long)m * SK + c;
        sj = s_saved[idx];
        float pj = (1.f - unimix) * sj + unimix / KD;
        t = gm[idx] / pj + __bfloat162float(gon[idx]) + (gon2 ? __bfloat162float(gon2[idx]) : 0.f);
      }
      float acc = t * sj;
#pragma unroll
      for (int off = 1; off < 64; off <<= 1) {
        if (off < KD) acc += __shfl_xor(acc, off, 64);
      }
      graw[m * KP + c] = __float2bfloat16(m < B ? (1.f - unimix) * sj * (t - acc) : 0.f);
    }
  }
  __syncthreads();
  // persist graw striped across the grid (all WGs hold the same image)
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < (long)B * SK;
       i += (long)gridDim.x * blockDim.x) {
    const int m = (int)(i / SK), c = (int)(i - (long)m * SK);
    graw_out[(long)m * gws + c] = graw[m * KP + c];
  }
  const int ncol0 = (blockIdx.x * 4 + wv) * 16;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  f32x4 acc = sc2_gemm_tile(graw, KP, W4t, SK, SK, P, ncol0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = kgrp * 4 + r;
    if (m < B && ncol0 + arow < P) gp_out[(long)m * P + ncol0 + arow] = __float2bfloat16(acc[r]);
  }
}

// ---------------------------------------------------------------------------
// scan2 backward LN(+SiLU) phase (phases 3 and 1): recompute the row-local
// LayerNorm backward into LDS per WG (no sync), then the next GEMM tile.
// Stripe WGs additionally persist gg and flush the LN affine grads.
// RESETS=false -> phase 3 (gr output, strided); RESETS=true -> phase 1
// (gx output with the episode-reset backward fused: carries + action grad +
// init-state accumulators).
// ---------------------------------------------------------------------------
template <bool RESETS>
__global__ void __launch_bounds__(256) scan2_blnsilu_kernel(
    const __hip_bfloat16* __restrict__ gy_in, long gys,  // grad wrt activated out [B, P]
    const __hip_bfloat16* __restrict__ g_in, long gis,   // saved pre-LN GEMM result
    const __hip_bfloat16* __restrict__ lnw, const __hip_bfloat16* __restrict__ lnb,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const __hip_bfloat16* __restrict__ Wt,               // [N, P] pre-transposed next-GEMM weight
    const __hip_bfloat16* __restrict__ f,                // [B] (RESETS)
    __hip_bfloat16* __restrict__ gg_out, long ggs,       // gg stripe persist
    float* __restrict__ glnw, float* __restrict__ glnb,  // fp32 accumulators
    __hip_bfloat16* __restrict__ out, long outs,         // gr_s[t] (phase 3) / unused (RESETS)
    __hip_bfloat16* __restrict__ gz_carry, float* __restrict__ giz_acc,
    __hip_bfloat16* __restrict__ ga_out,                 // [B, A] (RESETS)
    int B, int P, int N, int SK, int A) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int KP = P + 8;  // +8: LDS bank de-phasing (P % 64 == 0)
  __hip_bfloat16* gg = (__hip_bfloat16*)smem;                      // [16][P+8]
  float* s12 = (float*)(smem + ((16 * KP * 2 + 15) & ~15));        // [2][16]
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  // pass A (vectorized): per-row sums (wave wv owns rows 4wv..4wv+3) + gz
  for (int mr2 = 0; mr2 < 4; ++mr2) {
    const int m = wv * 4 + mr2;
    float s1 = 0.f, s2 = 0.f;
    if (m < B) {
      const float mn = mean[m], rs = rstd[m];
      for (int j0 = lane * 8; j0 < P; j0 += 512) {
        bf16x8 gv = *(const bf16x8*)(g_in + (long)m * gis + j0);
        bf16x8 gyv = *(const bf16x8*)(gy_in + (long)m * gys + j0);
        bf16x8 lwv = *(const bf16x8*)(lnw + j0);
        bf16x8 lbv = *(const bf16x8*)(lnb + j0);
        bf16x8 ogz;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float xh = ((float)gv[e] - mn) * rs;
          float w = (float)lwv[e];
          float z = xh * w + (float)lbv[e];
          float sig = 1.f / (1.f + expf(-z));
          float gz = (float)gyv[e] * sig * (1.f + z * (1.f - sig));
          ogz[e] = (__bf16)gz;
          float gxh = gz * w;
          s1 += gxh;
          s2 += gxh * xh;
        }
        *(bf16x8*)(gg + m * KP + j0) = ogz;  // temporarily gz
      }
      s1 = wave_sum(s1);
      s2 = wave_sum(s2);
    } else {
      bf16x8 zero{};
      for (int j0 = lane * 8; j0 < P; j0 += 512) *(bf16x8*)(gg + m * KP + j0) = zero;
    }
    if (lane == 0) {
      s12[m] = s1 / P;
      s12[16 + m] = s2 / P;
    }
  }
  __syncthreads();
  // LN affine grads: lane-per-column coalesced row loop, one stripe per WG
  for (int stripe = blockIdx.x; stripe < P / 64; stripe += gridDim.x) {
    const int j = stripe * 64 + (threadIdx.x & 63);
    if (threadIdx.x < 64) {
      float sw = 0.f, sb = 0.f;
      for (int m = 0; m < B; ++m) {
        float gz = __bfloat162float(gg[m * KP + j]);
        float xh = (__bfloat162float(g_in[(long)m * gis + j]) - mean[m]) * rstd[m];
        sw += gz * xh;
        sb += gz;
      }
      atomicAdd(&glnw[j], sw);
      atomicAdd(&glnb[j], sb);
    }
  }
  __syncthreads();
  // pass B: finalize gg in place (vectorized; skip the pad columns)
  for (int iv = threadIdx.x; iv < 16 * (P >> 3); iv += blockDim.x) {
    const int m = iv / (P >> 3), jv = iv - m * (P >> 3);
    if (m >= B) continue;
    const int j0 = jv << 3;
    bf16x8 gzv = *(const bf16x8*)(gg + m * KP + j0);
    bf16x8 gv = *(const bf16x8*)(g_in + (long)m * gis + j0);
    bf16x8 lwv = *(const bf16x8*)(lnw + j0);
    bf16x8 outv;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float xh = ((float)gv[e] - mean[m]) * rstd[m];
      outv[e] = (__bf16)(((float)gzv[e] * (float)lwv[e] - s12[m] - xh * s12[16 + m]) * rstd[m]);
    }
    *(bf16x8*)(gg + m * KP + j0) = outv;
  }
  __syncthreads();
  // persist gg stripes (for the batched weight-grad GEMMs), vectorized
  for (int stripe = blockIdx.x; stripe < P / 64; stripe += gridDim.x) {
    for (int i = threadIdx.x; i < B * 8; i += blockDim.x) {
      const int m = i / 8, jv = i & 7;
      const int j0 = stripe * 64 + jv * 8;
      *(bf16x8*)(gg_out + (long)m * ggs + j0) = *(const bf16x8*)(gg + m * KP + j0);
    }
  }
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int ncol0 = (blockIdx.x * 4 + wv) * 16;
  f32x4 acc = sc2_gemm_tile(gg, KP, Wt, P, P, N, ncol0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = kgrp * 4 + r;
    const int c = ncol0 + arow;
    if (m < B && c < N) {
      if (RESETS) {
        const float fb = __bfloat162float(f[m]);
        const float gv = acc[r];
        if (c < SK) {
          gz_carry[(long)m * SK + c] = __float2bfloat16((1.f - fb) * gv);
          giz_acc[(long)m * SK + c] += fb * gv;
        } else if (c < SK + A) {
          ga_out[(long)m * A + (c - SK)] = __float2bfloat16((1.f - fb) * gv);
        }
      } else {
        out[(long)m * outs + c] = __float2bfloat16(acc[r]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// scan2 backward phase 2: full Hafner-gate + LN backward recomputed per WG
// into LDS, then the ghu = gy @ W2t GEMM tile with the reset backward for the
// h columns fused into the epilogue.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256) scan2_bgru_kernel(
    const __hip_bfloat16* __restrict__ gh,        // g_h_seq[t] [B, H]
    const __hip_bfloat16* __restrict__ gh2,       // carry or null
    const __hip_bfloat16* __restrict__ gh3, long gh3s,  // gr_s[t][:, :H]
    const __hip_bfloat16* __restrict__ y_in, long ys2,  // saved pre-LN y [B, 3H]
    const __hip_bfloat16* __restrict__ hu, long hus,    // h' in [:, :H]
    const __hip_bfloat16* __restrict__ lnw, const __hip_bfloat16* __restrict__ lnb,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const __hip_bfloat16* __restrict__ W2t,       // [H+D, 3H]
    const __hip_bfloat16* __restrict__ f,         // [B]
    __hip_bfloat16* __restrict__ gy_out, long gys,      // gy_s[t]
    float* __restrict__ glnw, float* __restrict__ glnb,
    __hip_bfloat16* __restrict__ gh_carry,        // [B, H]
    float* __restrict__ gih_acc,                  // [B, H] fp32
    __hip_bfloat16* __restrict__ ghu_out,         // [B, H+D] scratch (u part consumed by b1)
    int B, int H, int D) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int DD = 3 * H;
  const int DDP = DD + 8;  // +8: LDS bank de-phasing
  __hip_bfloat16* gybuf = (__hip_bfloat16*)smem;                     // [16][3H+8]
  __hip_bfloat16* ghp = (__hip_bfloat16*)(smem + 16 * DDP * 2);      // [16][H]
  float* s12 = (float*)(smem + ((16 * DDP * 2 + 16 * H * 2 + 15) & ~15));  // [2][16]
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  // pass 1, wave per 4 rows, 8-wide vector loads: full Hafner-gate backward
  // into LDS (gz, ghp) + the LN row sums
  for (int mr2 = 0; mr2 < 4; ++mr2) {
    const int m = wv * 4 + mr2;
    float s1 = 0.f, s2 = 0.f;
    if (m < B) {
      const float mn = mean[m], rs = rstd[m];
      for (int j0 = lane * 8; j0 < H; j0 += 512) {
        bf16x8 yrv = *(const bf16x8*)(y_in + (long)m * ys2 + j0);
        bf16x8 ycv = *(const bf16x8*)(y_in + (long)m * ys2 + H + j0);
        bf16x8 yuv = *(const bf16x8*)(y_in + (long)m * ys2 + 2 * H + j0);
        bf16x8 ghv = *(const bf16x8*)(gh + (long)m * H + j0);
        bf16x8 gh3v = *(const bf16x8*)(gh3 + (long)m * gh3s + j0);
        bf16x8 huv = *(const bf16x8*)(hu + (long)m * hus + j0);
        bf16x8 lwr = *(const bf16x8*)(lnw + j0);
        bf16x8 lwc = *(const bf16x8*)(lnw + H + j0);
        bf16x8 lwu = *(const bf16x8*)(lnw + 2 * H + j0);
        bf16x8 lbr = *(const bf16x8*)(lnb + j0);
        bf16x8 lbc = *(const bf16x8*)(lnb + H + j0);
        bf16x8 lbu = *(const bf16x8*)(lnb + 2 * H + j0);
        bf16x8 gh2v{};
        if (gh2) gh2v = *(const bf16x8*)(gh2 + (long)m * H + j0);
        bf16x8 ozr, ozc, ozu, ophp;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float xh_r = ((float)yrv[e] - mn) * rs;
          float xh_c = ((float)ycv[e] - mn) * rs;
          float xh_u = ((float)yuv[e] - mn) * rs;
          float zr = xh_r * (float)lwr[e] + (float)lbr[e];
          float zc = xh_c * (float)lwc[e] + (float)lbc[e];
          float zu = xh_u * (float)lwu[e] + (float)lbu[e];
          float r = 1.f / (1.f + expf(-zr));
          float c = tanhf(r * zc);
          float u = 1.f / (1.f + expf(-(zu - 1.f)));
          float g = (float)ghv[e] + (gh2 ? (float)gh2v[e] : 0.f) + (float)gh3v[e];
          float gu = g * (c - (float)huv[e]);
          float gc = g * u;
          float gzu = gu * u * (1.f - u);
          float grc = gc * (1.f - c * c);
          float gzc = grc * r;
          float gr = grc * zc;
          float gzr = gr * r * (1.f - r);
          ophp[e] = (__bf16)(g * (1.f - u));
          ozr[e] = (__bf16)gzr;
          ozc[e] = (__bf16)gzc;
          ozu[e] = (__bf16)gzu;
          float gxh_r = gzr * (float)lwr[e];
          float gxh_c = gzc * (float)lwc[e];
          float gxh_u = gzu * (float)lwu[e];
          s1 += gxh_r + gxh_c + gxh_u;
          s2 += gxh_r * xh_r + gxh_c * xh_c + gxh_u * xh_u;
        }
        *(bf16x8*)(ghp + m * H + j0) = ophp;
        *(bf16x8*)(gybuf + m * DDP + j0) = ozr;
        *(bf16x8*)(gybuf + m * DDP + H + j0) = ozc;
        *(bf16x8*)(gybuf + m * DDP + 2 * H + j0) = ozu;
      }
      s1 = wave_sum(s1);
      s2 = wave_sum(s2);
    } else {
      bf16x8 zero{};
      for (int j0 = lane * 8; j0 < DD; j0 += 512) *(bf16x8*)(gybuf + m * DDP + j0) = zero;
    }
    if (lane == 0) {
      s12[m] = s1 / DD;
      s12[16 + m] = s2 / DD;
    }
  }
  __syncthreads();
  // LN affine grads: lane-per-column coalesced row loop, one stripe per WG
  for (int stripe = blockIdx.x; stripe < DD / 64; stripe += gridDim.x) {
    const int j = stripe * 64 + (threadIdx.x & 63);
    if (threadIdx.x < 64) {
      float sw = 0.f, sb = 0.f;
      for (int m = 0; m < B; ++m) {
        float gz = __bfloat162float(gybuf[m * DDP + j]);
        float xh = (__bfloat162float(y_in[(long)m * ys2 + j]) - mean[m]) * rstd[m];
        sw += gz * xh;
        sb += gz;
      }
      atomicAdd(&glnw[j], sw);
      atomicAdd(&glnb[j], sb);
    }
  }
  __syncthreads();
  // finalize gy in place (vectorized)
  for (int iv = threadIdx.x; iv < 16 * (DD >> 3); iv += blockDim.x) {
    const int m = iv / (DD >> 3), jv = iv - m * (DD >> 3);
    if (m >= B) continue;
    const int j0 = jv << 3;
    bf16x8 gzv = *(const bf16x8*)(gybuf + m * DDP + j0);
    bf16x8 yv = *(const bf16x8*)(y_in + (long)m * ys2 + j0);
    bf16x8 lwv = *(const bf16x8*)(lnw + j0);
    bf16x8 outv;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float xh = ((float)yv[e] - mean[m]) * rstd[m];
      outv[e] = (__bf16)(((float)gzv[e] * (float)lwv[e] - s12[m] - xh * s12[16 + m]) * rstd[m]);
    }
    *(bf16x8*)(gybuf + m * DDP + j0) = outv;
  }
  __syncthreads();
  for (int stripe = blockIdx.x; stripe < DD / 64; stripe += gridDim.x) {
    for (int i = threadIdx.x; i < B * 8; i += blockDim.x) {
      const int m = i / 8, jv = i & 7;
      const int j0 = stripe * 64 + jv * 8;
      *(bf16x8*)(gy_out + (long)m * gys + j0) = *(const bf16x8*)(gybuf + m * DDP + j0);
    }
  }
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int ncol0 = (blockIdx.x * 4 + wv) * 16;
  f32x4 acc = sc2_gemm_tile(gybuf, DDP, W2t, DD, DD, H + D, ncol0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = kgrp * 4 + r;
    const int c = ncol0 + arow;
    if (m < B && c < H + D) {
      if (c < H) {
        const float fb = __bfloat162float(f[m]);
        float g2 = acc[r] + __bfloat162float(ghp[m * H + c]);
        gh_carry[(long)m * H + c] = __float2bfloat16((1.f - fb) * g2);
        gih_acc[(long)m * H + c] += fb * g2;
      } else {
        ghu_out[(long)m * (H + D) + c] = __float2bfloat16(acc[r]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// scan3: split-K fused phase kernels (round-2 second pass).  The scan2
// kernels proved ~2-4x slower than hipblaslt's M=16 GEMMs in the real graph:
// with only N/64 workgroups the cold weight stream is latency-bound (too few
// outstanding loads chip-wide).  scan3 adopts hipblaslt-grade parallelism —
// grid = ntiles x KS workgroups, each computing one 16-col tile over a K
// slice (further quartered across its 4 waves), accumulated into an fp32
// scratch with device atomics.  Per tile, a generation ticket elects the
// LAST-arriving workgroup, which runs the epilogue (cat-ST head) or joins the
// second, ntiles-wide ticket round for the grid-spanning LayerNorm.
// Generation counting (arrivals compared against gen*KS / gen*ntiles) makes
// tickets monotonic — no reset races between the T graph-captured launches.
// ---------------------------------------------------------------------------

// wave-level combine of the 4 K-quarter partials through LDS; afterwards all
// four waves hold the full k-slice tile.
__device__ __forceinline__ f32x4 scan3_combine(f32x4 acc, float* comb) {
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int wv = threadIdx.x >> 6;
#pragma unroll
  for (int r = 0; r < 4; ++r) comb[(wv * 16 + kgrp * 4 + r) * 16 + arow] = acc[r];
  __syncthreads();
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = kgrp * 4 + r;
    acc[r] = comb[m * 16 + arow] + comb[(16 + m) * 16 + arow] + comb[(32 + m) * 16 + arow] +
             comb[(48 + m) * 16 + arow];
  }
  return acc;
}

// wave-0 accumulates the combined tile into the fp32 scratch; generation
// ticket elects the tile's last arriver (which gets an agent acquire).
__device__ __forceinline__ bool scan3_commit(const f32x4& acc, float* scratch, int N, int col0,
                                             int* tile_ticket, int arrivals) {
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int wv = threadIdx.x >> 6;
  if (wv == 0) {
#pragma unroll
    for (int r = 0; r < 4; ++r) atomicAdd(&scratch[(long)(kgrp * 4 + r) * N + col0 + arow], acc[r]);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  __shared__ int last_;
  if (threadIdx.x == 0) {
    int t = __hip_atomic_fetch_add(tile_ticket, 1, __ATOMIC_ACQ_REL, __HIP_MEMORY_SCOPE_AGENT);
    last_ = (t == arrivals - 1) ? 1 : 0;
    if (last_) __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  }
  __syncthreads();
  return last_ != 0;
}

// wave-0-only second round for the grid-spanning LayerNorm: publish this
// tile's partial sums, wait for every tile's last arriver, return row stats.
// mr_ is a 32-float LDS area ([0..15] mean, [16..31] rstd).
__device__ __forceinline__ void scan3_ln_round2(const f32x4& v, int N_total, int B, float* ws2,
                                                int* ticket2, int wait_count, float eps, float* mr_) {
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  float s = 0.f;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    float cs = sc2_colsum(v[r]);
    float cs2 = sc2_colsum(v[r] * v[r]);
    if (arow == 0) {
      atomicAdd(&ws2[kgrp * 4 + r], cs);
      atomicAdd(&ws2[16 + kgrp * 4 + r], cs2);
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  if (lane == 0) {
    __hip_atomic_fetch_add(ticket2, 1, __ATOMIC_ACQ_REL, __HIP_MEMORY_SCOPE_AGENT);
    int spins = 0;
    while (__hip_atomic_load(ticket2, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT) < wait_count &&
           spins < (1 << 27)) {
      __builtin_amdgcn_s_sleep(2);
      ++spins;
    }
  }
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  if (lane < 16) {
    float mean = ws2[lane] / N_total;
    float var = ws2[16 + lane] / N_total - mean * mean;
    mr_[lane] = mean;
    mr_[16 + lane] = rsqrtf(fmaxf(var, 0.f) + eps);
  }
}

// phases 1 & 3: C = A @ W^T + grid LayerNorm + SiLU.  RESETS assembles the
// episode-reset-masked x on the fly (z'/a') and persists x / h'.
template <bool RESETS, bool T0>
__global__ void __launch_bounds__(256) scan3_lnsilu_kernel(
    const __hip_bfloat16* __restrict__ a_in, long as_,
    const __hip_bfloat16* __restrict__ iz, const __hip_bfloat16* __restrict__ h_prev,
    const __hip_bfloat16* __restrict__ ih, const __hip_bfloat16* __restrict__ act,
    const __hip_bfloat16* __restrict__ f, const __hip_bfloat16* __restrict__ W,
    const __hip_bfloat16* __restrict__ lnw, const __hip_bfloat16* __restrict__ lnb,
    __hip_bfloat16* __restrict__ x_out, long xs, __hip_bfloat16* __restrict__ hu_out, long hus,
    __hip_bfloat16* __restrict__ g_out, long gs, float* __restrict__ mean_out,
    float* __restrict__ rstd_out, float* __restrict__ scratch, int* __restrict__ tickets,
    int* __restrict__ ticket2, float* __restrict__ ws2, int B, int SK, int A, int H, int N, int K,
    float eps, int hu_off, int gen) {
  __shared__ float comb[4 * 16 * 16];
  __shared__ float mr_[32];
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int wv = threadIdx.x >> 6;
  const int tile = blockIdx.x;
  const int KS = gridDim.y;
  const int col0 = tile * 16;
  const int KQ = (((K + KS - 1) / KS) + 31) & ~31;
  const int kbeg0 = min((int)blockIdx.y * KQ, K);
  const int kend0 = min(kbeg0 + KQ, K);
  const int KQW = (((kend0 - kbeg0 + 3) / 4) + 31) & ~31;
  const int kbeg = min(kbeg0 + wv * KQW, kend0);
  const int kend = min(kbeg + KQW, kend0);
  f32x4 acc;
  if (RESETS) {
    // on-the-fly reset-masked A: [z', a'] built per fragment load
    const int wrow = min(col0 + arow, N - 1);
    acc = (f32x4){0.f, 0.f, 0.f, 0.f};
    const int arow_c = arow < B ? arow : B - 1;
    const float fb = __bfloat162float(f[arow_c]);
    for (int k0 = kbeg; k0 < kend; k0 += 32) {
      const int k = k0 + kgrp * 8;
      bf16x8 a, b;
      if (k + 8 <= SK) {
        bf16x8 izv = *(const bf16x8*)(iz + (long)arow_c * SK + k);
        if (T0) {
#pragma unroll
          for (int e = 0; e < 8; ++e) a[e] = (__bf16)(fb * (float)izv[e]);
        } else {
          bf16x8 zpv = *(const bf16x8*)(a_in + (long)arow_c * SK + k);
#pragma unroll
          for (int e = 0; e < 8; ++e) a[e] = (__bf16)((1.f - fb) * (float)zpv[e] + fb * (float)izv[e]);
        }
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int kk = k + e;
          float v = 0.f;
          if (kk < SK) {
            float zp = T0 ? 0.f : __bfloat162float(a_in[(long)arow_c * SK + kk]);
            v = (1.f - fb) * zp + fb * __bfloat162float(iz[(long)arow_c * SK + kk]);
          } else if (kk < SK + A) {
            v = (1.f - fb) * __bfloat162float(act[(long)arow_c * A + (kk - SK)]);
          }
          a[e] = (__bf16)v;
        }
      }
      const __hip_bfloat16* p = W + (long)wrow * K + k;
      if (k + 8 <= K) {
        b = *(const bf16x8*)p;
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) b[e] = (k + e < K) ? (__bf16)__bfloat162float(p[e]) : (__bf16)0.f;
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    // persist x (K-striped by tile-0 slices) and h' (tile 0, slice 0)
    if (tile == 0) {
      for (int i = threadIdx.x; i < B * (kend0 - kbeg0); i += blockDim.x) {
        const int m = i / (kend0 - kbeg0), kk = kbeg0 + i % (kend0 - kbeg0);
        float v;
        if (kk < SK) {
          const float fm = __bfloat162float(f[m]);
          float zp = T0 ? 0.f : __bfloat162float(a_in[(long)m * SK + kk]);
          v = (1.f - fm) * zp + fm * __bfloat162float(iz[(long)m * SK + kk]);
        } else {
          const float fm = __bfloat162float(f[m]);
          v = (1.f - fm) * __bfloat162float(act[(long)m * A + (kk - SK)]);
        }
        x_out[(long)m * xs + kk] = __float2bfloat16(v);
      }
      if (blockIdx.y == 0) {
        for (int i = threadIdx.x; i < B * H; i += blockDim.x) {
          const int m = i / H, j = i - m * H;
          const float fm = __bfloat162float(f[m]);
          float hp = T0 ? 0.f : __bfloat162float(h_prev[(long)m * H + j]);
          st(hu_out, (long)m * hus + j, (1.f - fm) * hp + fm * __bfloat162float(ih[(long)m * H + j]));
        }
      }
    }
  } else {
    acc = sc2_gemm_tile_range_g(a_in, as_, B, W, K, K, N, col0, kbeg, kend);
  }
  acc = scan3_combine(acc, comb);
  if (!scan3_commit(acc, scratch, N, col0, &tickets[tile], gen * KS)) return;
  if (wv != 0) return;
  // wave-0 round 2: read the final tile, grid LayerNorm, epilogue, reset
  f32x4 v;
#pragma unroll
  for (int r = 0; r < 4; ++r) v[r] = scratch[(long)(kgrp * 4 + r) * N + col0 + arow];
  scan3_ln_round2(v, N, B, ws2, ticket2, gen * gridDim.x, eps, mr_);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = kgrp * 4 + r;
    const int j = col0 + arow;
    if (m < B) {
      g_out[(long)m * gs + j] = __float2bfloat16(v[r]);
      float z = (v[r] - mr_[m]) * mr_[16 + m] * __bfloat162float(lnw[j]) + __bfloat162float(lnb[j]);
      z = z / (1.f + __expf(-z));
      st(hu_out, (long)m * hus + hu_off + j, z);
    }
    scratch[(long)(kgrp * 4 + r) * N + col0 + arow] = 0.f;
  }
  if (tile == 0 && lane < 16 && lane < B) {
    mean_out[lane] = mr_[lane];
    rstd_out[lane] = mr_[16 + lane];
  }
}

// phase 2: the GRU projection (3 gate stripes per h column) + grid LayerNorm
// over 3H + Hafner gates.
__global__ void __launch_bounds__(256) scan3_gru_kernel(
    const __hip_bfloat16* __restrict__ hu, long hus, const __hip_bfloat16* __restrict__ W2,
    const __hip_bfloat16* __restrict__ lnw, const __hip_bfloat16* __restrict__ lnb,
    __hip_bfloat16* __restrict__ y_out, long ys2, __hip_bfloat16* __restrict__ h_out,
    __hip_bfloat16* __restrict__ h_out2, long h2s, float* __restrict__ mean_out,
    float* __restrict__ rstd_out, float* __restrict__ scratch, int* __restrict__ tickets,
    int* __restrict__ ticket2, float* __restrict__ ws2, int B, int H, int D, float eps, int gen) {
  __shared__ float comb[4 * 16 * 16];
  __shared__ float mr_[32];
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int wv = threadIdx.x >> 6;
  const int tile = blockIdx.x;
  const int KS = gridDim.y;
  const int K = H + D;
  const int col0 = tile * 16;
  const int KQ = (((K + KS - 1) / KS) + 31) & ~31;
  const int kbeg0 = min((int)blockIdx.y * KQ, K);
  const int kend0 = min(kbeg0 + KQ, K);
  const int KQW = (((kend0 - kbeg0 + 3) / 4) + 31) & ~31;
  const int kbeg = min(kbeg0 + wv * KQW, kend0);
  const int kend = min(kbeg + KQW, kend0);
  f32x4 accg[3];
#pragma unroll
  for (int g = 0; g < 3; ++g) {
    accg[g] = sc2_gemm_tile_range_g(hu, hus, B, W2, K, K, 3 * H, g * H + col0, kbeg, kend);
    accg[g] = scan3_combine(accg[g], comb);
    __syncthreads();
  }
  // commit all 3 stripes, one ticket
  if (wv == 0) {
#pragma unroll
    for (int g = 0; g < 3; ++g) {
#pragma unroll
      for (int r = 0; r < 4; ++r)
        atomicAdd(&scratch[(long)(kgrp * 4 + r) * 3 * H + g * H + col0 + arow], accg[g][r]);
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  __shared__ int last_;
  if (threadIdx.x == 0) {
    int t = __hip_atomic_fetch_add(&tickets[tile], 1, __ATOMIC_ACQ_REL, __HIP_MEMORY_SCOPE_AGENT);
    last_ = (t == gen * KS - 1) ? 1 : 0;
    if (last_) __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  }
  __syncthreads();
  if (!last_ || wv != 0) return;
  const int DD = 3 * H;
  f32x4 vg[3];
  float s1p = 0.f;
#pragma unroll
  for (int g = 0; g < 3; ++g) {
#pragma unroll
    for (int r = 0; r < 4; ++r) vg[g][r] = scratch[(long)(kgrp * 4 + r) * DD + g * H + col0 + arow];
  }
  // partial sums over the 3 stripes
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    float s = vg[0][r] + vg[1][r] + vg[2][r];
    float s2 = vg[0][r] * vg[0][r] + vg[1][r] * vg[1][r] + vg[2][r] * vg[2][r];
    float cs = sc2_colsum(s);
    float cs2 = sc2_colsum(s2);
    if (arow == 0) {
      atomicAdd(&ws2[kgrp * 4 + r], cs);
      atomicAdd(&ws2[16 + kgrp * 4 + r], cs2);
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  if (lane == 0) {
    __hip_atomic_fetch_add(ticket2, 1, __ATOMIC_ACQ_REL, __HIP_MEMORY_SCOPE_AGENT);
    int spins = 0;
    while (__hip_atomic_load(ticket2, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT) < gen * (int)gridDim.x &&
           spins < (1 << 27)) {
      __builtin_amdgcn_s_sleep(2);
      ++spins;
    }
  }
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  if (lane < 16) {
    float mean = ws2[lane] / DD;
    float var = ws2[16 + lane] / DD - mean * mean;
    mr_[lane] = mean;
    mr_[16 + lane] = rsqrtf(fmaxf(var, 0.f) + eps);
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = kgrp * 4 + r;
    const int j = col0 + arow;
    if (m < B) {
      // persist pre-LN y and run the Hafner gates
      y_out[(long)m * ys2 + j] = __float2bfloat16(vg[0][r]);
      y_out[(long)m * ys2 + H + j] = __float2bfloat16(vg[1][r]);
      y_out[(long)m * ys2 + 2 * H + j] = __float2bfloat16(vg[2][r]);
      const float mean = mr_[m], rstd = mr_[16 + m];
      float zr = (vg[0][r] - mean) * rstd * __bfloat162float(lnw[j]) + __bfloat162float(lnb[j]);
      float zc = (vg[1][r] - mean) * rstd * __bfloat162float(lnw[H + j]) + __bfloat162float(lnb[H + j]);
      float zu = (vg[2][r] - mean) * rstd * __bfloat162float(lnw[2 * H + j]) + __bfloat162float(lnb[2 * H + j]);
      float rg = 1.f / (1.f + expf(-zr));
      float cg = tanhf(rg * zc);
      float ug = 1.f / (1.f + expf(-(zu - 1.f)));
      float hp = __bfloat162float(hu[(long)m * hus + j]);
      float hv = ug * cg + (1.f - ug) * hp;
      h_out[(long)m * H + j] = __float2bfloat16(hv);
      h_out2[(long)m * h2s + j] = __float2bfloat16(hv);
    }
#pragma unroll
    for (int g = 0; g < 3; ++g) scratch[(long)(kgrp * 4 + r) * DD + g * H + col0 + arow] = 0.f;
  }
  if (tile == 0 && lane < 16 && lane < B) {
    mean_out[lane] = mr_[lane];
    rstd_out[lane] = mr_[16 + lane];
  }
}

// phase 4: raw = p @ W4^T + b4 + the unimix categorical-ST head.  Tiles are
// 64 columns wide (whole KD groups per epilogue owner); the 4 waves each own
// one 16-col subtile over the workgroup's K slice.
__global__ void __launch_bounds__(256) scan3_catst_kernel(
    const __hip_bfloat16* __restrict__ p_in, long ps, const __hip_bfloat16* __restrict__ W4,
    const __hip_bfloat16* __restrict__ b4, const float* __restrict__ urand,
    float* __restrict__ m_out, __hip_bfloat16* __restrict__ z_out, float* __restrict__ s_out,
    float* __restrict__ scratch, int* __restrict__ tickets, int B, int P, int SK, int KD,
    float unimix, int gen) {
  __shared__ float raw[16][72];
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int wv = threadIdx.x >> 6;
  const int tile = blockIdx.x;  // 64-col tile
  const int KS = gridDim.y;
  const int col0 = tile * 64 + wv * 16;
  const int KQ = (((P + KS - 1) / KS) + 31) & ~31;
  const int kbeg = min((int)blockIdx.y * KQ, P);
  const int kend = min(kbeg + KQ, P);
  f32x4 acc = sc2_gemm_tile_range_g(p_in, ps, B, W4, P, P, SK, col0, kbeg, kend);
#pragma unroll
  for (int r = 0; r < 4; ++r) atomicAdd(&scratch[(long)(kgrp * 4 + r) * SK + col0 + arow], acc[r]);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  __shared__ int last_;
  if (threadIdx.x == 0) {
    int t = __hip_atomic_fetch_add(&tickets[tile], 1, __ATOMIC_ACQ_REL, __HIP_MEMORY_SCOPE_AGENT);
    last_ = (t == gen * KS - 1) ? 1 : 0;
    if (last_) __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  }
  __syncthreads();
  if (!last_) return;
  for (int i = threadIdx.x; i < 16 * 64; i += blockDim.x) {
    const int m = i >> 6, c = i & 63;
    const long sidx = (long)m * SK + tile * 64 + c;
    raw[m][c] = scratch[sidx] + __bfloat162float(b4[tile * 64 + c]);
    scratch[sidx] = 0.f;
  }
  __syncthreads();
  const int gpw = 64 / KD;
  for (int task = threadIdx.x; task < 16 * gpw; task += blockDim.x) {
    const int m = task / gpw;
    if (m >= B) continue;
    const int gl = task - m * gpw;
    const float* row = &raw[m][gl * KD];
    const long gcol = (long)tile * 64 + gl * KD;
    float lmax = -1e30f;
    for (int j = 0; j < KD; ++j) lmax = fmaxf(lmax, row[j]);
    float lsum = 0.f;
    for (int j = 0; j < KD; ++j) lsum += expf(row[j] - lmax);
    const float inv = 1.f / lsum;
    const float* ur = urand + (long)m * SK + gcol;
    float* mro = m_out + (long)m * SK + gcol;
    float* sro = s_out + (long)m * SK + gcol;
    __hip_bfloat16* zro = z_out + (long)m * SK + gcol;
    float best = -1e30f;
    int best_j = 0;
    for (int j = 0; j < KD; ++j) {
      float sv = expf(row[j] - lmax) * inv;
      float pv = (1.f - unimix) * sv + unimix / KD;
      float mv = logf(pv);
      sro[j] = sv;
      mro[j] = mv;
      float tt = fmaxf(-logf(fmaxf(ur[j], 1e-20f)), 1e-20f);
      float score = mv - logf(tt);
      if (score > best) {
        best = score;
        best_j = j;
      }
    }
    for (int j = 0; j < KD; ++j) zro[j] = (__hip_bfloat16)(j == best_j ? 1.f : 0.f);
  }
}

// backward phase 4: per-slice categorical-ST backward into LDS, split-K
// gp = graw @ W4t into scratch, last arriver converts.
__global__ void __launch_bounds__(256) scan3_b4_kernel(
    const float* __restrict__ gm, const __hip_bfloat16* __restrict__ gon,
    const __hip_bfloat16* __restrict__ gon2, const float* __restrict__ s_saved,
    const __hip_bfloat16* __restrict__ W4t, __hip_bfloat16* __restrict__ graw_out, long gws,
    __hip_bfloat16* __restrict__ gp_out, float* __restrict__ scratch, int* __restrict__ tickets,
    int B, int SK, int P, int KD, float unimix, int gen) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int lane = threadIdx.x & 63;
  const int arow = lane & 15;
  const int kgrp = lane >> 4;
  const int wv = threadIdx.x >> 6;
  const int tile = blockIdx.x;
  const int KS = gridDim.y;
  const int col0 = tile * 16;
  // K slice of SK, aligned to max(KD, 32) so KD groups stay whole
  const int align = KD > 32 ? KD : 32;
  const int KQ = (((SK + KS - 1) / KS) + align - 1) & ~(align - 1);
  const int kbeg0 = min((int)blockIdx.y * KQ, SK);
  const int kend0 = min(kbeg0 + KQ, SK);
  const int KL = kend0 - kbeg0;
  const int KLP = ((KL + 31) & ~31) + 8;
  __hip_bfloat16* graw = (__hip_bfloat16*)smem;  // [16][KLP]
  float* comb = (float*)(smem + ((16 * KLP * 2 + 15) & ~15));
  // slice-local categorical-ST backward (wave-per-4-rows, coalesced)
  for (int mr2 = 0; mr2 < 4; ++mr2) {
    const int m = wv * 4 + mr2;
    for (int c0 = 0; c0 < KL; c0 += 64) {
      const int c = c0 + lane;
      float t = 0.f, sj = 0.f;
      if (m < B && c < KL) {
        const long idx = (long)m * SK + kbeg0 + c;
        sj = s_saved[idx];
        float pj = (1.f - unimix) * sj + unimix / KD;
        t = gm[idx] / pj + __bfloat162float(gon[idx]) + (gon2 ? __bfloat162float(gon2[idx]) : 0.f);
      }
      float acc = t * sj;
#pragma unroll
      for (int off = 1; off < 64; off <<= 1) {
        if (off < KD) acc += __shfl_xor(acc, off, 64);
      }
      if (c < KLP) graw[m * KLP + c] = __float2bfloat16((m < B && c < KL) ? (1.f - unimix) * sj * (t - acc) : 0.f);
    }
    // zero the 32-align padding tail
    for (int c = KL + lane; c < KLP; c += 64) graw[m * KLP + c] = (__hip_bfloat16)0.f;
  }
  __syncthreads();
  if (tile == 0) {
    for (int i = threadIdx.x; i < B * KL; i += blockDim.x) {
      const int m = i / KL, c = i - m * KL;
      graw_out[(long)m * gws + kbeg0 + c] = graw[m * KLP + c];
    }
  }
  // GEMM over the slice (waves split KL four ways from LDS)
  const int KQW = (((KL + 3) / 4) + 31) & ~31;
  const int kb = min(wv * KQW, KL);
  const int ke = min(kb + KQW, KL);
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  {
    const int wrow = min(col0 + arow, P - 1);
    for (int k0 = kb; k0 < ke; k0 += 32) {
      const int k = k0 + kgrp * 8;
      bf16x8 a = *(const bf16x8*)(graw + arow * KLP + k);
      bf16x8 b;
      const __hip_bfloat16* p = W4t + (long)wrow * SK + kbeg0 + k;
      if (kbeg0 + k + 8 <= SK && k + 8 <= ke) {
        b = *(const bf16x8*)p;
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e)
          b[e] = (k + e < ke) ? (__bf16)__bfloat162float(p[e]) : (__bf16)0.f;
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
  }
  acc = scan3_combine(acc, comb);
  if (!scan3_commit(acc, scratch, P, col0, &tickets[tile], gen * KS)) return;
  if (wv != 0) return;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int m = kgrp * 4 + r;
    const long sidx = (long)m * P + col0 + arow;
    if (m < B) gp_out[(long)m * P + col0 + arow] = __float2bfloat16(scratch[sidx]);
    scratch[sidx] = 0.f;
  }
}

// ---------------------------------------------------------------------------
// device-side replay gather (BASELINE north star):
// sequence windows are gathered by a HIP kernel reading the PINNED host ring
// directly over PCIe (zero-copy) into HBM on a side stream — replacing numpy
// fancy-indexing + host->device copies (reference buffers.py:493-511).
// ptrs[s] = host base address of sample s's ring array for this key;
// out[l, s, :] = ring[(starts[s] + l) % cap, :]  (time-major, matching the
// [n_samples, L, batch, ...] layout sample_tensors produces).
// ---------------------------------------------------------------------------

template <int UNIT>
__global__ void replay_gather_kernel(const long* __restrict__ ptrs, const long* __restrict__ starts,
                                     unsigned char* __restrict__ out, long S, long L, long cap,
                                     long row_bytes) {
  const long row_units = row_bytes / UNIT;
  const long total = S * L * row_units;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long u = i % row_units;
    const long sl = i / row_units;
    const long s = sl % S;
    const long l = sl / S;
    const long src_row = (starts[s] + l) % cap;
    const unsigned char* src = (const unsigned char*)ptrs[s] + src_row * row_bytes + u * UNIT;
    unsigned char* dst = out + (l * S + s) * row_bytes + u * UNIT;
    if (UNIT == 16)
      *(uint4*)dst = *(const uint4*)src;
    else if (UNIT == 4)
      *(unsigned int*)dst = *(const unsigned int*)src;
    else
      *dst = *src;
  }
}

void replay_gather(const torch::Tensor& ptrs, const torch::Tensor& starts, torch::Tensor out, long L,
                   long cap, long row_bytes) {
  TORCH_CHECK(ptrs.is_cuda() && starts.is_cuda() && out.is_cuda() && out.is_contiguous());
  TORCH_CHECK(ptrs.scalar_type() == at::kLong && starts.scalar_type() == at::kLong);
  const long S = ptrs.numel();
  TORCH_CHECK((long)out.numel() * out.element_size() == S * L * row_bytes, "replay_gather size");
  const int unit = (row_bytes % 16 == 0) ? 16 : (row_bytes % 4 == 0 ? 4 : 1);
  const long total = S * L * (row_bytes / unit);
  int blocks = (int)std::min((total + kBlock - 1) / kBlock, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (unit == 16)
    hipLaunchKernelGGL(replay_gather_kernel<16>, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       ptrs.data_ptr<long>(), starts.data_ptr<long>(), (unsigned char*)out.data_ptr(),
                       S, L, cap, row_bytes);
  else if (unit == 4)
    hipLaunchKernelGGL(replay_gather_kernel<4>, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       ptrs.data_ptr<long>(), starts.data_ptr<long>(), (unsigned char*)out.data_ptr(),
                       S, L, cap, row_bytes);
  else
    hipLaunchKernelGGL(replay_gather_kernel<1>, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       ptrs.data_ptr<long>(), starts.data_ptr<long>(), (unsigned char*)out.data_ptr(),
                       S, L, cap, row_bytes);
}

// ---------------------------------------------------------------------------
// fused LSTM cell gates (ppo_recurrent): the
// reference steps nn.LSTM per timestep in a Python loop; here the input
// projection batches over T outside and each step is one gates kernel.
// torch gate order i|f|g|o:  c' = σ(f)c + σ(i)tanh(g);  h = σ(o)tanh(c').
// The episode-reset mask is folded in: c_used = (1-first)*c_prev.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void lstm_gates_fwd_kernel(const T* __restrict__ y, const T* __restrict__ c_prev,
                                      const T* __restrict__ first, T* __restrict__ h_out,
                                      T* __restrict__ c_out, long N, int H) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < N * (long)H;
       i += (long)gridDim.x * blockDim.x) {
    const long n = i / H;
    const int j = (int)(i - n * H);
    const T* yr = y + n * 4 * (long)H;
    float ig = 1.f / (1.f + expf(-ld(yr, j)));
    float fg = 1.f / (1.f + expf(-ld(yr, H + j)));
    float gg = tanhf(ld(yr, 2 * H + j));
    float og = 1.f / (1.f + expf(-ld(yr, 3 * H + j)));
    float cp = ld(c_prev, i) * (first ? (1.f - ld(first, n)) : 1.f);
    float c = fg * cp + ig * gg;
    st(c_out, i, c);
    st(h_out, i, og * tanhf(c));
  }
}

template <typename T>
__global__ void lstm_gates_bwd_kernel(const T* __restrict__ gh, const T* __restrict__ gh2,
                                      const T* __restrict__ first2, const T* __restrict__ gc_carry,
                                      const T* __restrict__ y, const T* __restrict__ c_prev,
                                      const T* __restrict__ c_out, const T* __restrict__ first,
                                      T* __restrict__ gy, T* __restrict__ gc_prev, long N, int H) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < N * (long)H;
       i += (long)gridDim.x * blockDim.x) {
    const long n = i / H;
    const int j = (int)(i - n * H);
    const T* yr = y + n * 4 * (long)H;
    float ig = 1.f / (1.f + expf(-ld(yr, j)));
    float fg = 1.f / (1.f + expf(-ld(yr, H + j)));
    float gg = tanhf(ld(yr, 2 * H + j));
    float og = 1.f / (1.f + expf(-ld(yr, 3 * H + j)));
    float mask = first ? (1.f - ld(first, n)) : 1.f;
    float cp = ld(c_prev, i) * mask;
    float c = ld(c_out, i);
    float tc = tanhf(c);
    float g = ld(gh, i) + (gh2 ? ld(gh2, i) * (first2 ? (1.f - ld(first2, n)) : 1.f) : 0.f);
    float gc = g * og * (1.f - tc * tc) + (gc_carry ? ld(gc_carry, i) : 0.f);
    float go = g * tc;
    float gi = gc * gg;
    float gf = gc * cp;
    float gg_ = gc * ig;
    T* gyr = gy + n * 4 * (long)H;
    st(gyr, j, gi * ig * (1.f - ig));
    st(gyr, H + j, gf * fg * (1.f - fg));
    st(gyr, 2 * H + j, gg_ * (1.f - gg * gg));
    st(gyr, 3 * H + j, go * og * (1.f - og));
    st(gc_prev, i, gc * fg * mask);
  }
}

void lstm_gates_fwd(const torch::Tensor& y, const torch::Tensor& c_prev,
                    const c10::optional<torch::Tensor>& first, torch::Tensor h_out,
                    torch::Tensor c_out) {
  CHECK_IN(y);
  TORCH_CHECK(c_prev.is_contiguous() && h_out.is_contiguous() && c_out.is_contiguous());
  long N = c_prev.size(0) * (c_prev.dim() > 2 ? c_prev.size(1) : 1);
  N = c_prev.numel() / c_prev.size(-1);
  int H = (int)c_prev.size(-1);
  long n = N * H;
  int blocks = (int)std::min((n + kBlock - 1) / kBlock, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, y.scalar_type(), "lstm_gates_fwd", [&] {
    using T = scalar_t;
    const T* fp = first.has_value() ? (const T*)first->data_ptr() : nullptr;
    hipLaunchKernelGGL(lstm_gates_fwd_kernel<T>, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       (const T*)y.data_ptr(), (const T*)c_prev.data_ptr(), fp, (T*)h_out.data_ptr(),
                       (T*)c_out.data_ptr(), N, H);
  });
}

void lstm_gates_bwd(const torch::Tensor& gh, const c10::optional<torch::Tensor>& gh2,
                    const c10::optional<torch::Tensor>& first2,
                    const c10::optional<torch::Tensor>& gc_carry, const torch::Tensor& y,
                    const torch::Tensor& c_prev, const torch::Tensor& c_out,
                    const c10::optional<torch::Tensor>& first, torch::Tensor gy,
                    torch::Tensor gc_prev) {
  CHECK_IN(y);
  TORCH_CHECK(gy.is_contiguous() && gc_prev.is_contiguous());
  long N = c_prev.numel() / c_prev.size(-1);
  int H = (int)c_prev.size(-1);
  long n = N * H;
  int blocks = (int)std::min((n + kBlock - 1) / kBlock, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, y.scalar_type(), "lstm_gates_bwd", [&] {
    using T = scalar_t;
    const T* fp = first.has_value() ? (const T*)first->data_ptr() : nullptr;
    const T* g2 = gh2.has_value() ? (const T*)gh2->data_ptr() : nullptr;
    const T* f2 = first2.has_value() ? (const T*)first2->data_ptr() : nullptr;
    const T* gcc = gc_carry.has_value() ? (const T*)gc_carry->data_ptr() : nullptr;
    hipLaunchKernelGGL(lstm_gates_bwd_kernel<T>, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       (const T*)gh.data_ptr(), g2, f2, gcc, (const T*)y.data_ptr(),
                       (const T*)c_prev.data_ptr(), (const T*)c_out.data_ptr(), fp,
                       (T*)gy.data_ptr(), (T*)gc_prev.data_ptr(), N, H);
  });
}

// ---------------------------------------------------------------------------
// scan3 host wrappers
// ---------------------------------------------------------------------------

static inline const __hip_bfloat16* s3_bp(const torch::Tensor& t) {
  return (const __hip_bfloat16*)t.data_ptr();
}
static inline __hip_bfloat16* s3_bpm(torch::Tensor& t) { return (__hip_bfloat16*)t.data_ptr(); }

static inline int s3_ks(int ntiles, int K) {
  // target 96-256 workgroups, K-slices of >= ~128
  int ks = 192 / ntiles;
  ks = std::max(1, std::min(8, ks));
  while (ks > 1 && (K + ks - 1) / ks < 96) --ks;
  return ks;
}

void scan3_f1(const c10::optional<torch::Tensor>& z_prev, const torch::Tensor& iz,
              const c10::optional<torch::Tensor>& h_prev, const torch::Tensor& ih,
              const torch::Tensor& act, const torch::Tensor& f, const torch::Tensor& W1,
              const torch::Tensor& lnw, const torch::Tensor& lnb, torch::Tensor x_out,
              torch::Tensor hu_out, torch::Tensor g_out, torch::Tensor mean, torch::Tensor rstd,
              torch::Tensor scratch, torch::Tensor tickets, torch::Tensor ticket2, torch::Tensor ws2,
              double eps, long gen) {
  const int B = (int)act.size(0), A = (int)act.size(1), SK = (int)iz.size(1), H = (int)ih.size(1);
  const int N = (int)W1.size(0), K = (int)W1.size(1);
  TORCH_CHECK(B <= 16 && N % 16 == 0 && K == SK + A && SK % 8 == 0 && W1.is_contiguous());
  const int ntiles = N / 16;
  TORCH_CHECK(scratch.numel() >= 16 * (long)N && tickets.numel() >= ntiles);
  const int KS = s3_ks(ntiles, K);
  auto stream = at::cuda::getCurrentCUDAStream();
  const bool t0 = !z_prev.has_value();
  const __hip_bfloat16* zp = t0 ? nullptr : s3_bp(*z_prev);
  const __hip_bfloat16* hp = h_prev.has_value() ? s3_bp(*h_prev) : nullptr;
#define S3_F1(T0V)                                                                                     \
  hipLaunchKernelGGL((scan3_lnsilu_kernel<true, T0V>), dim3(ntiles, KS), dim3(256), 0,                 \
                     stream.stream(), zp, 0, s3_bp(iz), hp, s3_bp(ih), s3_bp(act), s3_bp(f),           \
                     s3_bp(W1), s3_bp(lnw), s3_bp(lnb), s3_bpm(x_out), x_out.stride(0),                \
                     s3_bpm(hu_out), hu_out.stride(0), s3_bpm(g_out), g_out.stride(0),                 \
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), scratch.data_ptr<float>(),        \
                     tickets.data_ptr<int>(), ticket2.data_ptr<int>(), ws2.data_ptr<float>(), B, SK,   \
                     A, H, N, K, (float)eps, H, (int)gen)
  if (t0)
    S3_F1(true);
  else
    S3_F1(false);
#undef S3_F1
}

void scan3_f3(const torch::Tensor& a_in, const torch::Tensor& W3, const torch::Tensor& lnw,
              const torch::Tensor& lnb, torch::Tensor p_out, torch::Tensor g_out, torch::Tensor mean,
              torch::Tensor rstd, torch::Tensor scratch, torch::Tensor tickets, torch::Tensor ticket2,
              torch::Tensor ws2, double eps, long gen) {
  const int B = (int)a_in.size(0);
  const int N = (int)W3.size(0), K = (int)W3.size(1);
  TORCH_CHECK(B <= 16 && N % 16 == 0 && a_in.size(1) == K && W3.is_contiguous());
  TORCH_CHECK((K % 8) == 0 && (a_in.stride(0) % 8) == 0, "scan3_f3: 16B-aligned A rows");
  const int ntiles = N / 16;
  const int KS = s3_ks(ntiles, K);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL((scan3_lnsilu_kernel<false, false>), dim3(ntiles, KS), dim3(256), 0,
                     stream.stream(), s3_bp(a_in), a_in.stride(0), nullptr, nullptr, nullptr, nullptr,
                     nullptr, s3_bp(W3), s3_bp(lnw), s3_bp(lnb), nullptr, 0, s3_bpm(p_out),
                     p_out.stride(0), s3_bpm(g_out), g_out.stride(0), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), scratch.data_ptr<float>(), tickets.data_ptr<int>(),
                     ticket2.data_ptr<int>(), ws2.data_ptr<float>(), B, 0, 0, 0, N, K, (float)eps, 0,
                     (int)gen);
}

void scan3_f2(const torch::Tensor& hu, const torch::Tensor& W2, const torch::Tensor& lnw,
              const torch::Tensor& lnb, torch::Tensor y_out, torch::Tensor h_out, torch::Tensor h_out2,
              torch::Tensor mean, torch::Tensor rstd, torch::Tensor scratch, torch::Tensor tickets,
              torch::Tensor ticket2, torch::Tensor ws2, double eps, long gen) {
  const int B = (int)hu.size(0), K = (int)hu.size(1);
  const int H = (int)h_out.size(1), D = K - H;
  TORCH_CHECK(B <= 16 && H % 16 == 0 && W2.size(0) == 3 * H && W2.size(1) == K && W2.is_contiguous());
  TORCH_CHECK(h_out.is_contiguous() && (hu.stride(0) % 8) == 0 && (K % 8) == 0);
  const int ntiles = H / 16;
  TORCH_CHECK(scratch.numel() >= 16 * 3 * (long)H && tickets.numel() >= ntiles);
  const int KS = s3_ks(ntiles, K);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(scan3_gru_kernel, dim3(ntiles, KS), dim3(256), 0, stream.stream(), s3_bp(hu),
                     hu.stride(0), s3_bp(W2), s3_bp(lnw), s3_bp(lnb), s3_bpm(y_out), y_out.stride(0),
                     s3_bpm(h_out), s3_bpm(h_out2), h_out2.stride(0), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), scratch.data_ptr<float>(), tickets.data_ptr<int>(),
                     ticket2.data_ptr<int>(), ws2.data_ptr<float>(), B, H, D, (float)eps, (int)gen);
}

void scan3_f4(const torch::Tensor& p_in, const torch::Tensor& W4, const torch::Tensor& b4,
              const torch::Tensor& urand, torch::Tensor m_out, torch::Tensor z_out, torch::Tensor s_out,
              torch::Tensor scratch, torch::Tensor tickets, long KD, double unimix, long gen) {
  const int B = (int)p_in.size(0), P = (int)p_in.size(1);
  const int SK = (int)W4.size(0);
  TORCH_CHECK(B <= 16 && SK % 64 == 0 && W4.size(1) == P && W4.is_contiguous());
  TORCH_CHECK(KD <= 64 && 64 % KD == 0 && m_out.is_contiguous() && z_out.is_contiguous() && s_out.is_contiguous());
  TORCH_CHECK((P % 8) == 0 && (p_in.stride(0) % 8) == 0);
  const int ntiles = SK / 64;
  TORCH_CHECK(scratch.numel() >= 16 * (long)SK && tickets.numel() >= ntiles);
  const int KS = s3_ks(ntiles, P);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(scan3_catst_kernel, dim3(ntiles, KS), dim3(256), 0, stream.stream(), s3_bp(p_in),
                     p_in.stride(0), s3_bp(W4), s3_bp(b4), urand.data_ptr<float>(),
                     m_out.data_ptr<float>(), s3_bpm(z_out), s_out.data_ptr<float>(),
                     scratch.data_ptr<float>(), tickets.data_ptr<int>(), B, P, SK, (int)KD,
                     (float)unimix, (int)gen);
}

void scan3_b4(const torch::Tensor& gm, const torch::Tensor& gon, const c10::optional<torch::Tensor>& gon2,
              const torch::Tensor& s_saved, const torch::Tensor& W4t, torch::Tensor graw_out,
              torch::Tensor gp_out, torch::Tensor scratch, torch::Tensor tickets, long KD,
              double unimix, long gen) {
  const int B = (int)gon.size(0), SK = (int)gon.size(1);
  const int P = (int)W4t.size(0);
  TORCH_CHECK(B <= 16 && SK % 64 == 0 && P % 16 == 0 && W4t.size(1) == SK && W4t.is_contiguous());
  TORCH_CHECK(KD <= 64 && (KD & (KD - 1)) == 0 && gp_out.is_contiguous());
  const int ntiles = P / 16;
  TORCH_CHECK(scratch.numel() >= 16 * (long)P && tickets.numel() >= ntiles);
  const int KS = s3_ks(ntiles, SK);
  const int align = KD > 32 ? (int)KD : 32;
  const int KQ = (((SK + KS - 1) / KS) + align - 1) & ~(align - 1);
  const int KLP = ((KQ + 31) & ~31) + 8;
  const size_t shmem = ((16 * KLP * 2 + 15) & ~15) + 4 * 16 * 16 * sizeof(float);
  auto stream = at::cuda::getCurrentCUDAStream();
  const __hip_bfloat16* g2 = gon2.has_value() ? s3_bp(*gon2) : nullptr;
  hipLaunchKernelGGL(scan3_b4_kernel, dim3(ntiles, KS), dim3(256), shmem, stream.stream(),
                     gm.data_ptr<float>(), s3_bp(gon), g2, s_saved.data_ptr<float>(), s3_bp(W4t),
                     s3_bpm(graw_out), graw_out.stride(0), s3_bpm(gp_out), scratch.data_ptr<float>(),
                     tickets.data_ptr<int>(), B, SK, P, (int)KD, (float)unimix, (int)gen);
}

// ---------------------------------------------------------------------------
// tiled 2-D transpose (64x64 LDS tiles, padded rows): torch's .t().contiguous()
// on bf16 weights is an uncoalesced 2-byte strided copy; the scan2 backward
// transposes ~10 MB of weights per step and needs this to be bandwidth-bound.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void transpose2d_kernel(const T* __restrict__ in, T* __restrict__ out, int R, int C) {
  __shared__ T tile[64][72];  // +8 pad: conflict-free transposed reads
  const int tr = blockIdx.y * 64;
  const int tc = blockIdx.x * 64;
  for (int i = threadIdx.x; i < 64 * 64; i += blockDim.x) {
    const int r = i >> 6, c = i & 63;
    T v = (T)0;
    if (tr + r < R && tc + c < C) v = in[(long)(tr + r) * C + tc + c];
    tile[r][c] = v;
  }
  __syncthreads();
  for (int i = threadIdx.x; i < 64 * 64; i += blockDim.x) {
    const int r2 = i >> 6, c2 = i & 63;  // output row tile = input cols
    if (tc + r2 < C && tr + c2 < R) out[(long)(tc + r2) * R + tr + c2] = tile[c2][r2];
  }
}

torch::Tensor transpose2d(const torch::Tensor& in) {
  CHECK_IN(in);
  TORCH_CHECK(in.dim() == 2);
  const int R = (int)in.size(0), C = (int)in.size(1);
  auto out = torch::empty({C, R}, in.options());
  dim3 grid((C + 63) / 64, (R + 63) / 64);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, in.scalar_type(), "transpose2d", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL(transpose2d_kernel<T>, grid, dim3(256), 0, stream.stream(),
                       (const T*)in.data_ptr(), (T*)out.data_ptr(), R, C);
  });
  return out;
}

// ---------------------------------------------------------------------------
// fused behaviour-learning losses (the DV3 actor/critic loss sections,
// sheeprl/algos/dreamer_v3/dreamer_v3.py:262-325): each is ONE kernel per
// direction instead of the ~30-launch autograd elementwise chain.
// ---------------------------------------------------------------------------

// mean of a TwoHotEncodingDistribution: symexp(sum softmax(logits) * bins)
// (inference-only: used where the value is consumed detached)
template <typename T>
__global__ void twohot_mean_kernel(const T* __restrict__ logits, float* __restrict__ out, long N, int K,
                                   float low, float high) {
  const int lane = threadIdx.x & 63;
  const long row = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= N) return;
  const T* lr = logits + row * (long)K;
  float mx = -1e30f;
  for (int j = lane; j < K; j += 64) mx = fmaxf(mx, ld(lr, j));
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) mx = fmaxf(mx, __shfl_xor(mx, off, 64));
  const float step = (high - low) / (K - 1);
  float se = 0.f, dot = 0.f;
  for (int j = lane; j < K; j += 64) {
    float e = expf(ld(lr, j) - mx);
    se += e;
    dot += e * (low + j * step);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    se += __shfl_xor(se, off, 64);
    dot += __shfl_xor(dot, off, 64);
  }
  if (lane == 0) {
    float m = dot / se;
    out[row] = (m >= 0.f ? 1.f : -1.f) * (expf(fabsf(m)) - 1.f);  // symexp
  }
}

torch::Tensor twohot_mean(const torch::Tensor& logits, double low, double high) {
  CHECK_IN(logits);
  int K = (int)logits.size(-1);
  long N = logits.numel() / K;
  auto out = torch::empty({N}, logits.options().dtype(at::kFloat));
  const int rpb = kBlock / 64;
  int blocks = (int)((N + rpb - 1) / rpb);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, logits.scalar_type(), "twohot_mean", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL(twohot_mean_kernel<T>, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       (const T*)logits.data_ptr(), out.data_ptr<float>(), N, K, (float)low, (float)high);
  });
  return out;
}

// REINFORCE actor loss (discrete, single head):
//   loss = -(1/(HZ*F)) sum_{t<HZ,f} disc * (sum_a m*act*adv - ent_coef * sum_a e^m m)
// m are the fused categorical head's NORMALIZED log-probs.
template <typename T>
__global__ void reinforce_fwd_kernel(const float* __restrict__ m, const T* __restrict__ act,
                                     const float* __restrict__ adv, const float* __restrict__ disc,
                                     float* __restrict__ out, long HZF, int A, float ent_coef) {
  __shared__ float lds[9];
  float acc = 0.f;
  for (long r = blockIdx.x * (long)blockDim.x + threadIdx.x; r < HZF;
       r += (long)gridDim.x * blockDim.x) {
    const float* mr = m + r * A;
    const T* ar = act + r * A;
    float logp = 0.f, ent = 0.f;
    for (int a = 0; a < A; ++a) {
      float mv = mr[a];
      logp += mv * ld(ar, a);
      ent -= expf(mv) * mv;
    }
    acc += disc[r] * (logp * adv[r] + ent_coef * ent);
  }
  acc = block_sum(acc, lds);
  if (threadIdx.x == 0) atomicAdd(out, -acc / HZF);
}

template <typename T>
__global__ void reinforce_bwd_kernel(const float* __restrict__ g, const float* __restrict__ m,
                                     const T* __restrict__ act, const float* __restrict__ adv,
                                     const float* __restrict__ disc, float* __restrict__ gm, long HZF,
                                     long TOT, int A, float ent_coef) {
  const float gs = g[0];
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < TOT * A;
       i += (long)gridDim.x * blockDim.x) {
    const long r = i / A;
    if (r >= HZF) {
      gm[i] = 0.f;  // the t = HZ row contributes nothing
      continue;
    }
    float mv = m[i];
    float dent = -expf(mv) * (1.f + mv);
    gm[i] = gs * (-1.f / HZF) * disc[r] * (ld(act, i) * adv[r] + ent_coef * dent);
  }
}

std::vector<torch::Tensor> reinforce_fwd(const torch::Tensor& m, const torch::Tensor& act,
                                         const torch::Tensor& adv, const torch::Tensor& disc, long HZF,
                                         double ent_coef) {
  CHECK_IN(m);
  TORCH_CHECK(m.scalar_type() == at::kFloat && adv.is_contiguous() && disc.is_contiguous());
  int A = (int)m.size(-1);
  auto out = torch::zeros({}, m.options());
  int blocks = (int)std::min((HZF + kBlock - 1) / kBlock, (long)240);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, act.scalar_type(), "reinforce_fwd", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL(reinforce_fwd_kernel<T>, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       m.data_ptr<float>(), (const T*)act.data_ptr(), adv.data_ptr<float>(),
                       disc.data_ptr<float>(), out.data_ptr<float>(), HZF, A, (float)ent_coef);
  });
  return {out};
}

torch::Tensor reinforce_bwd(const torch::Tensor& g, const torch::Tensor& m, const torch::Tensor& act,
                            const torch::Tensor& adv, const torch::Tensor& disc, long HZF,
                            double ent_coef) {
  int A = (int)m.size(-1);
  long TOT = m.numel() / A;
  auto gm = torch::empty_like(m);
  int blocks = (int)std::min((TOT * A + kBlock - 1) / kBlock, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, act.scalar_type(), "reinforce_bwd", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL(reinforce_bwd_kernel<T>, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       g.data_ptr<float>(), m.data_ptr<float>(), (const T*)act.data_ptr(),
                       adv.data_ptr<float>(), disc.data_ptr<float>(), gm.data_ptr<float>(), HZF, TOT, A,
                       (float)ent_coef);
  });
  return gm;
}

// critic loss: mean_r disc[r] * (-lp(t1[r]) - lp(t2[r])) over shared logits
// (two two-hot cross-entropies per row in one pass; bwd recomputes softmax)
__global__ void vloss2_fwd_kernel(const float* __restrict__ logits, const float* __restrict__ t1,
                                  const float* __restrict__ t2, const float* __restrict__ disc,
                                  float* __restrict__ out, float* __restrict__ lse_out, long N, int K,
                                  float low, float high) {
  __shared__ float lds[9];
  const int lane = threadIdx.x & 63;
  const long wave0 = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const long nwaves = (long)gridDim.x * (blockDim.x >> 6);
  float acc = 0.f;
  for (long row = wave0; row < N; row += nwaves) {
    const float* lr = logits + row * (long)K;
    float mx = -1e30f;
    for (int j = lane; j < K; j += 64) mx = fmaxf(mx, lr[j]);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) mx = fmaxf(mx, __shfl_xor(mx, off, 64));
    float se = 0.f;
    for (int j = lane; j < K; j += 64) se += expf(lr[j] - mx);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) se += __shfl_xor(se, off, 64);
    const float lse = mx + logf(se);
    if (lane == 0) {
      int lo1, hi1, lo2, hi2;
      float wl1, wh1, wl2, wh2;
      twohot_idx(t1[row], low, high, K, lo1, hi1, wl1, wh1);
      twohot_idx(t2[row], low, high, K, lo2, hi2, wl2, wh2);
      float lp1 = wl1 * (lr[lo1] - lse) + wh1 * (lr[hi1] - lse);
      float lp2 = wl2 * (lr[lo2] - lse) + wh2 * (lr[hi2] - lse);
      lse_out[row] = lse;
      acc += disc[row] * (-(lp1)-lp2) / N;
    }
  }
  // one atomic per block instead of one per row (a single fp32 address
  // saturates at ~88 atomics/us)
  acc = block_sum(acc, lds);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

__global__ void vloss2_bwd_kernel(const float* __restrict__ g, const float* __restrict__ logits,
                                  const float* __restrict__ t1, const float* __restrict__ t2,
                                  const float* __restrict__ disc, const float* __restrict__ lse,
                                  float* __restrict__ gl, long N, int K, float low, float high) {
  const float gs = g[0];
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < N * (long)K;
       i += (long)gridDim.x * blockDim.x) {
    const long row = i / K;
    const int j = (int)(i - row * K);
    int lo1, hi1, lo2, hi2;
    float wl1, wh1, wl2, wh2;
    twohot_idx(t1[row], low, high, K, lo1, hi1, wl1, wh1);
    twohot_idx(t2[row], low, high, K, lo2, hi2, wl2, wh2);
    const float tw = (j == lo1 ? wl1 : 0.f) + (j == hi1 ? wh1 : 0.f) + (j == lo2 ? wl2 : 0.f) +
                     (j == hi2 ? wh2 : 0.f);
    gl[i] = gs * disc[row] * (2.f * expf(logits[i] - lse[row]) - tw) / N;
  }
}

std::vector<torch::Tensor> vloss2_fwd(const torch::Tensor& logits, const torch::Tensor& t1,
                                      const torch::Tensor& t2, const torch::Tensor& disc, double low,
                                      double high) {
  CHECK_IN(logits);
  TORCH_CHECK(logits.scalar_type() == at::kFloat && t1.is_contiguous() && t2.is_contiguous() &&
              disc.is_contiguous());
  int K = (int)logits.size(-1);
  long N = logits.numel() / K;
  auto out = torch::zeros({}, logits.options());
  auto lse = torch::empty({N}, logits.options());
  const int rpb = kBlock / 64;
  int blocks = (int)std::min((N + rpb - 1) / rpb, (long)480);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(vloss2_fwd_kernel, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                     logits.data_ptr<float>(), t1.data_ptr<float>(), t2.data_ptr<float>(),
                     disc.data_ptr<float>(), out.data_ptr<float>(), lse.data_ptr<float>(), N, K,
                     (float)low, (float)high);
  return {out, lse};
}

torch::Tensor vloss2_bwd(const torch::Tensor& g, const torch::Tensor& logits, const torch::Tensor& t1,
                         const torch::Tensor& t2, const torch::Tensor& disc, const torch::Tensor& lse,
                         double low, double high) {
  int K = (int)logits.size(-1);
  long N = logits.numel() / K;
  auto gl = torch::empty_like(logits);
  int blocks = (int)std::min((N * K + kBlock - 1) / kBlock, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(vloss2_bwd_kernel, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                     g.data_ptr<float>(), logits.data_ptr<float>(), t1.data_ptr<float>(),
                     t2.data_ptr<float>(), disc.data_ptr<float>(), lse.data_ptr<float>(),
                     gl.data_ptr<float>(), N, K, (float)low, (float)high);
  return gl;
}

// ---------------------------------------------------------------------------
// fused reconstruction-loss NLL kernels (SURVEY.md §2.8 item 6 extension):
// one reduction launch forward + one elementwise launch backward instead of
// the autograd sub/pow/sum/cast chains of MSEDistribution / SymlogDistribution
// / Bernoulli log_prob (sheeprl/utils/distribution.py:152-221 semantics).
//   MODE 0: log_prob = -sum_d (pred - tgt)^2
//   MODE 1: log_prob = -sum_d (pred - symlog(tgt))^2
//   MODE 2: log_prob =  sum_d (tgt * pred - softplus(pred))   (Bernoulli logits)
// ---------------------------------------------------------------------------

template <typename T, int MODE>
__device__ __forceinline__ float nll_elem(float p, float t) {
  if (MODE == 0) {
    float d = p - t;
    return -d * d;
  } else if (MODE == 1) {
    float ts = copysignf(logf(fabsf(t) + 1.f), t);
    float d = p - ts;
    return -(p - ts) * d;
  }
  return t * p - (fmaxf(p, 0.f) + log1pf(expf(-fabsf(p))));
}

template <typename T, int MODE>
__global__ void nll_fwd_kernel(const T* __restrict__ pred, const float* __restrict__ tgt,
                               float* __restrict__ out, long R, long D) {
  const int lane = threadIdx.x & 63;
  const long wave = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const long nwaves = (long)gridDim.x * (blockDim.x >> 6);
  constexpr int V = 16 / sizeof(T);  // pred vector width (8 bf16 / 4 fp32)
  for (long r = wave; r < R; r += nwaves) {
    const T* pr = pred + r * D;
    const float* tr = tgt + r * D;
    float acc = 0.f;
    if ((D & (V - 1)) == 0) {
      for (long j0 = (long)lane * V; j0 < D; j0 += 64 * V) {
        LnVec<T, V> pv;
        pv.u = *(const uint4*)(pr + j0);
        float tv[V];
#pragma unroll
        for (int e = 0; e < V; ++e) tv[e] = tr[j0 + e];
#pragma unroll
        for (int e = 0; e < V; ++e) acc += nll_elem<T, MODE>(ld(pv.e, e), tv[e]);
      }
    } else {
      for (long j = lane; j < D; j += 64) acc += nll_elem<T, MODE>(ld(pr, j), tr[j]);
    }
    acc = wave_sum(acc);
    if (lane == 0) out[r] = acc;
  }
}

template <typename T, int MODE>
__global__ void nll_bwd_kernel(const float* __restrict__ g, const T* __restrict__ pred,
                               const float* __restrict__ tgt, T* __restrict__ gpred, long R, long D) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < R * D;
       i += (long)gridDim.x * blockDim.x) {
    const long r = i / D;
    float p = ld(pred, i), t = tgt[i];
    float gr = g[r];
    float gv;
    if (MODE == 0) {
      gv = gr * -2.f * (p - t);
    } else if (MODE == 1) {
      float ts = copysignf(logf(fabsf(t) + 1.f), t);
      gv = gr * -2.f * (p - ts);
    } else {
      gv = gr * (t - 1.f / (1.f + expf(-p)));
    }
    st(gpred, i, gv);
  }
}

torch::Tensor nll_fwd(const torch::Tensor& pred, const torch::Tensor& tgt, long D, long mode) {
  CHECK_IN(pred);
  TORCH_CHECK(tgt.is_contiguous() && tgt.scalar_type() == at::kFloat && tgt.numel() == pred.numel());
  long R = pred.numel() / D;
  auto out = torch::empty({R}, pred.options().dtype(at::kFloat));
  // enough waves to cover the big image-reconstruction rows with slack
  int blocks = (int)std::min(std::max((R + 3) / 4, (R * D + 32767) / 32768), (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, pred.scalar_type(), "nll_fwd", [&] {
    using T = scalar_t;
    switch (mode) {
      case 0:
        hipLaunchKernelGGL((nll_fwd_kernel<T, 0>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                           (const T*)pred.data_ptr(), tgt.data_ptr<float>(), out.data_ptr<float>(), R, D);
        break;
      case 1:
        hipLaunchKernelGGL((nll_fwd_kernel<T, 1>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                           (const T*)pred.data_ptr(), tgt.data_ptr<float>(), out.data_ptr<float>(), R, D);
        break;
      default:
        hipLaunchKernelGGL((nll_fwd_kernel<T, 2>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                           (const T*)pred.data_ptr(), tgt.data_ptr<float>(), out.data_ptr<float>(), R, D);
    }
  });
  return out;
}

torch::Tensor nll_bwd(const torch::Tensor& g, const torch::Tensor& pred, const torch::Tensor& tgt,
                      long D, long mode) {
  CHECK_IN(pred);
  TORCH_CHECK(g.is_contiguous() && g.scalar_type() == at::kFloat);
  long R = pred.numel() / D;
  auto gpred = torch::empty_like(pred);
  long n = R * D;
  int blocks = (int)std::min((n + kBlock - 1) / kBlock, (long)2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, pred.scalar_type(), "nll_bwd", [&] {
    using T = scalar_t;
    switch (mode) {
      case 0:
        hipLaunchKernelGGL((nll_bwd_kernel<T, 0>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                           g.data_ptr<float>(), (const T*)pred.data_ptr(), tgt.data_ptr<float>(),
                           (T*)gpred.data_ptr(), R, D);
        break;
      case 1:
        hipLaunchKernelGGL((nll_bwd_kernel<T, 1>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                           g.data_ptr<float>(), (const T*)pred.data_ptr(), tgt.data_ptr<float>(),
                           (T*)gpred.data_ptr(), R, D);
        break;
      default:
        hipLaunchKernelGGL((nll_bwd_kernel<T, 2>), dim3(blocks), dim3(kBlock), 0, stream.stream(),
                           g.data_ptr<float>(), (const T*)pred.data_ptr(), tgt.data_ptr<float>(),
                           (T*)gpred.data_ptr(), R, D);
    }
  });
  return gpred;
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

static inline const __hip_bfloat16* sc2_bp(const torch::Tensor& t) {
  return (const __hip_bfloat16*)t.data_ptr();
}
static inline __hip_bfloat16* sc2_bpm(torch::Tensor& t) { return (__hip_bfloat16*)t.data_ptr(); }

void scan2_f1(const c10::optional<torch::Tensor>& z_prev, const torch::Tensor& iz,
              const c10::optional<torch::Tensor>& h_prev, const torch::Tensor& ih,
              const torch::Tensor& act, const torch::Tensor& f, const torch::Tensor& W1,
              const torch::Tensor& lnw, const torch::Tensor& lnb, torch::Tensor x_out,
              torch::Tensor hu_out, torch::Tensor g_out, torch::Tensor mean, torch::Tensor rstd,
              torch::Tensor ws2, torch::Tensor ticket, double eps) {
  const int B = (int)act.size(0), A = (int)act.size(1), SK = (int)iz.size(1), H = (int)ih.size(1);
  const int N = (int)W1.size(0), K = (int)W1.size(1);
  TORCH_CHECK(B <= 16 && N % 64 == 0 && K == SK + A && SK % 8 == 0 && W1.is_contiguous());
  const int KP = ((K + 31) & ~31) + 8;
  const size_t shmem = ((16 * KP * 2 + 15) & ~15) + 32 * sizeof(float);
  auto stream = at::cuda::getCurrentCUDAStream();
  const bool t0 = !z_prev.has_value();
  const __hip_bfloat16* zp = t0 ? nullptr : sc2_bp(*z_prev);
  const __hip_bfloat16* hp = h_prev.has_value() ? sc2_bp(*h_prev) : nullptr;
#define SC2_F1_LAUNCH(T0V)                                                                            \
  hipLaunchKernelGGL((scan2_lnsilu_kernel<true, T0V, false>), dim3(N / 64), dim3(256), shmem,         \
                     stream.stream(),                                                                 \
                     zp, 0, sc2_bp(iz), hp, sc2_bp(ih), sc2_bp(act), sc2_bp(f), sc2_bp(W1),           \
                     sc2_bp(lnw), sc2_bp(lnb), sc2_bpm(x_out), x_out.stride(0), sc2_bpm(hu_out),      \
                     hu_out.stride(0), sc2_bpm(g_out), g_out.stride(0), mean.data_ptr<float>(),       \
                     rstd.data_ptr<float>(), ws2.data_ptr<float>(), ticket.data_ptr<int>(), B, SK, A, \
                     H, N, K, (float)eps, H)
  if (t0)
    SC2_F1_LAUNCH(true);
  else
    SC2_F1_LAUNCH(false);
#undef SC2_F1_LAUNCH
}

void scan2_f3(const torch::Tensor& a_in, const torch::Tensor& W3, const torch::Tensor& lnw,
              const torch::Tensor& lnb, torch::Tensor p_out, torch::Tensor g_out, torch::Tensor mean,
              torch::Tensor rstd, torch::Tensor ws2, torch::Tensor ticket, double eps) {
  const int B = (int)a_in.size(0);
  const int N = (int)W3.size(0), K = (int)W3.size(1);
  TORCH_CHECK(B <= 16 && N % 64 == 0 && a_in.size(1) == K && W3.is_contiguous());
  const int KP = ((K + 31) & ~31) + 8;
  // long-K shapes (the representation GEMM, K = H+E ~ 4.6k at S): K-split the
  // 4 waves over ONE 16-col tile over GLOBAL A and run N/16 workgroups — 4x
  // the weight-stream concurrency, no LDS staging
  const bool ksplit = K >= 3072 && N / 16 <= 256 && (K % 8) == 0 && (a_in.stride(0) % 8) == 0;
  const size_t shmem = ksplit ? (4096 + 32 * sizeof(float))
                              : ((16 * KP * 2 + 15) & ~15) + 32 * sizeof(float);
  TORCH_CHECK(shmem <= 160 * 1024, "scan2_f3: K too large for LDS");
  auto stream = at::cuda::getCurrentCUDAStream();
  if (ksplit)
    hipLaunchKernelGGL((scan2_lnsilu_kernel<false, false, true>), dim3(N / 16), dim3(256), shmem,
                       stream.stream(),
                       sc2_bp(a_in), a_in.stride(0), nullptr, nullptr, nullptr, nullptr, nullptr,
                       sc2_bp(W3), sc2_bp(lnw), sc2_bp(lnb), nullptr, 0, sc2_bpm(p_out), p_out.stride(0),
                       sc2_bpm(g_out), g_out.stride(0), mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       ws2.data_ptr<float>(), ticket.data_ptr<int>(), B, 0, 0, 0, N, K, (float)eps, 0);
  else
    hipLaunchKernelGGL((scan2_lnsilu_kernel<false, false, false>), dim3(N / 64), dim3(256), shmem,
                       stream.stream(),
                       sc2_bp(a_in), a_in.stride(0), nullptr, nullptr, nullptr, nullptr, nullptr,
                       sc2_bp(W3), sc2_bp(lnw), sc2_bp(lnb), nullptr, 0, sc2_bpm(p_out), p_out.stride(0),
                       sc2_bpm(g_out), g_out.stride(0), mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       ws2.data_ptr<float>(), ticket.data_ptr<int>(), B, 0, 0, 0, N, K, (float)eps, 0);
}

void scan2_f2(const torch::Tensor& hu, const torch::Tensor& W2, const torch::Tensor& lnw,
              const torch::Tensor& lnb, torch::Tensor y_out, torch::Tensor h_out, torch::Tensor h_out2,
              torch::Tensor mean, torch::Tensor rstd, torch::Tensor ws2, torch::Tensor ticket,
              double eps) {
  const int B = (int)hu.size(0), K = (int)hu.size(1);
  const int H = (int)h_out.size(1), D = K - H;
  TORCH_CHECK(B <= 16 && H % 64 == 0 && W2.size(0) == 3 * H && W2.size(1) == K && W2.is_contiguous());
  TORCH_CHECK(h_out.is_contiguous());
  const int KP = ((K + 31) & ~31) + 8;
  const size_t shmem = ((16 * KP * 2 + 15) & ~15) + 32 * sizeof(float);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(scan2_gru_kernel, dim3(H / 64), dim3(256), shmem, stream.stream(), sc2_bp(hu),
                     hu.stride(0), sc2_bp(W2), sc2_bp(lnw), sc2_bp(lnb), sc2_bpm(y_out), y_out.stride(0),
                     sc2_bpm(h_out), sc2_bpm(h_out2), h_out2.stride(0), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), ws2.data_ptr<float>(), ticket.data_ptr<int>(), B, H, D,
                     (float)eps);
}

void scan2_f4(const torch::Tensor& p_in, const torch::Tensor& W4, const torch::Tensor& b4,
              const torch::Tensor& urand, torch::Tensor m_out, torch::Tensor z_out, torch::Tensor s_out,
              long KD, double unimix) {
  const int B = (int)p_in.size(0), P = (int)p_in.size(1);
  const int SK = (int)W4.size(0);
  TORCH_CHECK(B <= 16 && SK % 64 == 0 && W4.size(1) == P && W4.is_contiguous());
  TORCH_CHECK(KD <= 64 && 64 % KD == 0 && m_out.is_contiguous() && z_out.is_contiguous() && s_out.is_contiguous());
  const int KP = ((P + 31) & ~31) + 8;
  const size_t shmem = ((16 * KP * 2 + 15) & ~15) + 16 * 64 * sizeof(float);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(scan2_catst_kernel, dim3(SK / 64), dim3(256), shmem, stream.stream(), sc2_bp(p_in),
                     p_in.stride(0), sc2_bp(W4), sc2_bp(b4), urand.data_ptr<float>(),
                     m_out.data_ptr<float>(), sc2_bpm(z_out), s_out.data_ptr<float>(), B, P, SK, (int)KD,
                     (float)unimix);
}

void scan2_b4(const torch::Tensor& gm, const torch::Tensor& gon, const c10::optional<torch::Tensor>& gon2,
              const torch::Tensor& s_saved, const torch::Tensor& W4t, torch::Tensor graw_out,
              torch::Tensor gp_out, long KD, double unimix) {
  const int B = (int)gon.size(0), SK = (int)gon.size(1);
  const int P = (int)W4t.size(0);
  TORCH_CHECK(B <= 16 && SK % 64 == 0 && P % 64 == 0 && W4t.size(1) == SK && W4t.is_contiguous());
  TORCH_CHECK(KD <= 64 && (KD & (KD - 1)) == 0 && gp_out.is_contiguous());
  const size_t shmem = 16 * (SK + 8) * 2;
  auto stream = at::cuda::getCurrentCUDAStream();
  const __hip_bfloat16* g2 = gon2.has_value() ? sc2_bp(*gon2) : nullptr;
  hipLaunchKernelGGL(scan2_b4_kernel, dim3(P / 64), dim3(256), shmem, stream.stream(),
                     gm.data_ptr<float>(), sc2_bp(gon), g2, s_saved.data_ptr<float>(), sc2_bp(W4t),
                     sc2_bpm(graw_out), graw_out.stride(0), sc2_bpm(gp_out), B, SK, P, (int)KD,
                     (float)unimix);
}

void scan2_b3(const torch::Tensor& gy_in, const torch::Tensor& g_in, const torch::Tensor& lnw,
              const torch::Tensor& lnb, const torch::Tensor& mean, const torch::Tensor& rstd,
              const torch::Tensor& Wt, torch::Tensor gg_out, torch::Tensor glnw, torch::Tensor glnb,
              torch::Tensor out) {
  const int B = (int)gy_in.size(0), P = (int)gy_in.size(1);
  const int N = (int)Wt.size(0);
  TORCH_CHECK(B <= 16 && P % 64 == 0 && Wt.size(1) == P && Wt.is_contiguous());
  TORCH_CHECK(gy_in.stride(0) % 8 == 0 && g_in.stride(0) % 8 == 0 && gg_out.stride(0) % 8 == 0,
              "scan2 blnsilu: 16B-aligned rows required");
  const size_t shmem = ((16 * (P + 8) * 2 + 15) & ~15) + 32 * sizeof(float);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL((scan2_blnsilu_kernel<false>), dim3((N + 63) / 64), dim3(256), shmem, stream.stream(),
                     sc2_bp(gy_in), gy_in.stride(0), sc2_bp(g_in), g_in.stride(0), sc2_bp(lnw),
                     sc2_bp(lnb), mean.data_ptr<float>(), rstd.data_ptr<float>(), sc2_bp(Wt), nullptr,
                     sc2_bpm(gg_out), gg_out.stride(0), glnw.data_ptr<float>(), glnb.data_ptr<float>(),
                     sc2_bpm(out), out.stride(0), nullptr, nullptr, nullptr, B, P, N, 0, 0);
}

void scan2_b1(const torch::Tensor& gy_in, const torch::Tensor& g_in, const torch::Tensor& lnw,
              const torch::Tensor& lnb, const torch::Tensor& mean, const torch::Tensor& rstd,
              const torch::Tensor& Wt, const torch::Tensor& f, torch::Tensor gg_out, torch::Tensor glnw,
              torch::Tensor glnb, torch::Tensor gz_carry, torch::Tensor giz_acc, torch::Tensor ga_out) {
  const int B = (int)gy_in.size(0), P = (int)gy_in.size(1);
  const int N = (int)Wt.size(0);
  const int SK = (int)gz_carry.size(1), A = (int)ga_out.size(1);
  TORCH_CHECK(B <= 16 && P % 64 == 0 && Wt.size(1) == P && Wt.is_contiguous() && N == SK + A);
  const size_t shmem = ((16 * P * 2 + 15) & ~15) + 32 * sizeof(float);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL((scan2_blnsilu_kernel<true>), dim3((N + 63) / 64), dim3(256), shmem, stream.stream(),
                     sc2_bp(gy_in), gy_in.stride(0), sc2_bp(g_in), g_in.stride(0), sc2_bp(lnw),
                     sc2_bp(lnb), mean.data_ptr<float>(), rstd.data_ptr<float>(), sc2_bp(Wt), sc2_bp(f),
                     sc2_bpm(gg_out), gg_out.stride(0), glnw.data_ptr<float>(), glnb.data_ptr<float>(),
                     nullptr, 0, sc2_bpm(gz_carry), giz_acc.data_ptr<float>(), sc2_bpm(ga_out), B, P, N,
                     SK, A);
}

void scan2_b2(const torch::Tensor& gh, const c10::optional<torch::Tensor>& gh2, const torch::Tensor& gh3,
              const torch::Tensor& y_in, const torch::Tensor& hu, const torch::Tensor& lnw,
              const torch::Tensor& lnb, const torch::Tensor& mean, const torch::Tensor& rstd,
              const torch::Tensor& W2t, const torch::Tensor& f, torch::Tensor gy_out, torch::Tensor glnw,
              torch::Tensor glnb, torch::Tensor gh_carry, torch::Tensor gih_acc, torch::Tensor ghu_out) {
  const int B = (int)gh.size(0), H = (int)gh.size(1);
  const int HD = (int)W2t.size(0), D = HD - H;
  TORCH_CHECK(B <= 16 && H % 64 == 0 && HD % 64 == 0 && W2t.size(1) == 3 * H && W2t.is_contiguous());
  TORCH_CHECK(gh.is_contiguous() && gh_carry.is_contiguous() && ghu_out.is_contiguous());
  TORCH_CHECK(gh3.stride(0) % 8 == 0 && y_in.stride(0) % 8 == 0 && hu.stride(0) % 8 == 0 &&
              gy_out.stride(0) % 8 == 0, "scan2_b2: 16B-aligned rows required");
  const size_t shmem = ((16 * (3 * H + 8) * 2 + 16 * H * 2 + 15) & ~15) + 32 * sizeof(float);
  TORCH_CHECK(shmem <= 160 * 1024, "scan2_b2: H too large for LDS");
  auto stream = at::cuda::getCurrentCUDAStream();
  const __hip_bfloat16* g2 = gh2.has_value() ? sc2_bp(*gh2) : nullptr;
  hipLaunchKernelGGL(scan2_bgru_kernel, dim3(HD / 64), dim3(256), shmem, stream.stream(), sc2_bp(gh), g2,
                     sc2_bp(gh3), gh3.stride(0), sc2_bp(y_in), y_in.stride(0), sc2_bp(hu), hu.stride(0),
                     sc2_bp(lnw), sc2_bp(lnb), mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     sc2_bp(W2t), sc2_bp(f), sc2_bpm(gy_out), gy_out.stride(0), glnw.data_ptr<float>(),
                     glnb.data_ptr<float>(), sc2_bpm(gh_carry), gih_acc.data_ptr<float>(),
                     sc2_bpm(ghu_out), B, H, D);
}

// ---------------------------------------------------------------------------
// Moments percentile-EMA update (parity:
// sheeprl/algos/dreamer_v3/utils.py:56-63).  ONE kernel replaces the
// torch.quantile sort path (multiple launches) plus the six elementwise
// EMA/clamp launches: single workgroup loads the gathered returns into LDS,
// bitonic-sorts them, and thread 0 computes the two linearly-interpolated
// quantiles (torch.quantile "linear" semantics), EMA-updates the low/high
// buffers in place (fixed storage — hipGraph-replayable) and writes
// invscale = max(high-low, 1/max).  n <= 32768 (128 KB LDS); the DV3 shape
// is H*B*T = 15*1024 = 15360 per rank.
__global__ void moments_update_kernel(const float* __restrict__ x, int n, int npad,
                                      float* __restrict__ low, float* __restrict__ high,
                                      float* __restrict__ invscale, float p_low, float p_high,
                                      float decay, float inv_max) {
  extern __shared__ float sm[];
  for (int i = threadIdx.x; i < npad; i += blockDim.x) sm[i] = (i < n) ? x[i] : INFINITY;
  __syncthreads();
  for (int k = 2; k <= npad; k <<= 1) {
    for (int j = k >> 1; j > 0; j >>= 1) {
      for (int i = threadIdx.x; i < npad; i += blockDim.x) {
        const int ixj = i ^ j;
        if (ixj > i) {
          const bool up = ((i & k) == 0);
          const float a = sm[i], b = sm[ixj];
          if (up ? (a > b) : (a < b)) {
            sm[i] = b;
            sm[ixj] = a;
          }
        }
      }
      __syncthreads();
    }
  }
  if (threadIdx.x == 0) {
    const float posl = p_low * (float)(n - 1);
    const float posh = p_high * (float)(n - 1);
    const int ll = (int)floorf(posl), hl = (int)floorf(posh);
    const int lh = min(ll + 1, n - 1), hh = min(hl + 1, n - 1);
    const float ql = sm[ll] + (posl - (float)ll) * (sm[lh] - sm[ll]);
    const float qh = sm[hl] + (posh - (float)hl) * (sm[hh] - sm[hl]);
    const float L = low[0] * decay + (1.f - decay) * ql;
    const float H = high[0] * decay + (1.f - decay) * qh;
    low[0] = L;
    high[0] = H;
    invscale[0] = fmaxf(H - L, inv_max);
  }
}

torch::Tensor moments_update(const torch::Tensor& x, torch::Tensor low, torch::Tensor high,
                             double p_low, double p_high, double decay, double max_) {
  CHECK_IN(x);
  TORCH_CHECK(x.scalar_type() == at::kFloat && low.scalar_type() == at::kFloat &&
              high.scalar_type() == at::kFloat);
  const long n = x.numel();
  TORCH_CHECK(n >= 1 && n <= 32768, "moments_update: n out of range (LDS sort cap)");
  int npad = 1;
  while (npad < n) npad <<= 1;
  auto invscale = torch::empty({}, x.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(moments_update_kernel, dim3(1), dim3(1024), (size_t)npad * sizeof(float),
                     stream.stream(), x.data_ptr<float>(), (int)n, npad, low.data_ptr<float>(),
                     high.data_ptr<float>(), invscale.data_ptr<float>(), (float)p_low, (float)p_high,
                     (float)decay, (float)(1.0 / max_));
  return invscale;
}

// ---------------------------------------------------------------------------
// Fused tanh-Normal sample + summed log-prob (the SAC actor head;
// SURVEY.md §2.8 item 13, parity: sheeprl/algos/sac/agent.py:123-142).
// Forward folds std = exp(clamp(logstd)), the reparameterized sample,
// tanh squash, action rescale and the per-row log-prob sum
//   logp = sum_a [ logN(x) - log(1 - tanh(x)^2) - log(action_scale) ]
// into one kernel (replaces ~12 eager launches); backward recomputes
// std/x/y from the saved inputs in one kernel (replaces ~20).
// Rows are processed one per thread (A is small: action dims).
template <typename T>
__global__ void tanh_normal_fwd_kernel(const float* __restrict__ mean, const float* __restrict__ logstd,
                                       const float* __restrict__ eps, const float* __restrict__ scale,
                                       const float* __restrict__ bias, T* __restrict__ action,
                                       float* __restrict__ logp, long B, int A, float lmin, float lmax) {
  for (long r = blockIdx.x * (long)blockDim.x + threadIdx.x; r < B; r += (long)gridDim.x * blockDim.x) {
    const long base = r * A;
    float acc = 0.f;
    for (int a = 0; a < A; ++a) {
      const float ls = fminf(fmaxf(logstd[base + a], lmin), lmax);
      const float sd = __expf(ls);
      const float e = eps[base + a];
      const float xv = mean[base + a] + sd * e;
      const float y = tanhf(xv);
      action[base + a] = (T)(y * scale[a] + bias[a]);
      // -2*(log2 - x - softplus(-2x)) == log(1-y^2); softplus via the
      // overflow-safe max+log1p form
      const float z = -2.f * xv;
      const float sp = fmaxf(z, 0.f) + log1pf(__expf(-fabsf(z)));
      acc += -0.5f * e * e - ls - 0.91893853320467274f - 2.f * (0.69314718055994531f - xv - sp) -
             __logf(scale[a]);
    }
    logp[r] = acc;
  }
}

template <typename T>
__global__ void tanh_normal_bwd_kernel(const T* __restrict__ gaction, const float* __restrict__ glogp,
                                       const float* __restrict__ mean, const float* __restrict__ logstd,
                                       const float* __restrict__ eps, const float* __restrict__ scale,
                                       float* __restrict__ dmean, float* __restrict__ dlogstd, long B,
                                       int A, float lmin, float lmax) {
  for (long r = blockIdx.x * (long)blockDim.x + threadIdx.x; r < B; r += (long)gridDim.x * blockDim.x) {
    const long base = r * A;
    const float glp = glogp[r];
    for (int a = 0; a < A; ++a) {
      const float ls0 = logstd[base + a];
      const float ls = fminf(fmaxf(ls0, lmin), lmax);
      const float sd = __expf(ls);
      const float e = eps[base + a];
      const float xv = mean[base + a] + sd * e;
      const float y = tanhf(xv);
      const float dy = (float)gaction[base + a] * scale[a];
      // d logp / dx = 2*tanh(x); dy/dx = 1 - y^2
      const float dx = dy * (1.f - y * y) + 2.f * y * glp;
      dmean[base + a] = dx;
      const float dsd = dx * e - glp / sd;
      // clamp passes gradient on the closed interval (torch semantics)
      dlogstd[base + a] = (ls0 >= lmin && ls0 <= lmax) ? dsd * sd : 0.f;
    }
  }
}

std::vector<torch::Tensor> tanh_normal_fwd(const torch::Tensor& mean, const torch::Tensor& logstd,
                                           const torch::Tensor& eps, const torch::Tensor& scale,
                                           const torch::Tensor& bias, const torch::Tensor& like,
                                           double lmin, double lmax) {
  CHECK_IN(mean);
  CHECK_IN(logstd);
  CHECK_IN(eps);
  CHECK_IN(scale);
  CHECK_IN(bias);
  TORCH_CHECK(mean.scalar_type() == at::kFloat && logstd.scalar_type() == at::kFloat &&
              eps.scalar_type() == at::kFloat && scale.scalar_type() == at::kFloat);
  const int A = (int)scale.numel();
  const long B = mean.numel() / A;
  TORCH_CHECK((long)A * B == mean.numel() && mean.sizes() == logstd.sizes() && mean.sizes() == eps.sizes());
  auto action = torch::empty_like(mean, mean.options().dtype(like.scalar_type()));
  auto sizes = mean.sizes().vec();
  sizes.back() = 1;
  auto logp = torch::empty(sizes, mean.options());
  const int blocks = (int)((B + kBlock - 1) / kBlock);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, action.scalar_type(), "tanh_normal_fwd", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL(tanh_normal_fwd_kernel<T>, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       mean.data_ptr<float>(), logstd.data_ptr<float>(), eps.data_ptr<float>(),
                       scale.data_ptr<float>(), bias.data_ptr<float>(), (T*)action.data_ptr(),
                       logp.data_ptr<float>(), B, A, (float)lmin, (float)lmax);
  });
  return {action, logp};
}

std::vector<torch::Tensor> tanh_normal_bwd(const torch::Tensor& gaction, const torch::Tensor& glogp,
                                           const torch::Tensor& mean, const torch::Tensor& logstd,
                                           const torch::Tensor& eps, const torch::Tensor& scale,
                                           double lmin, double lmax) {
  CHECK_IN(gaction);
  CHECK_IN(glogp);
  CHECK_IN(mean);
  const int A = (int)scale.numel();
  const long B = mean.numel() / A;
  auto dmean = torch::empty_like(mean);
  auto dlogstd = torch::empty_like(logstd);
  const int blocks = (int)((B + kBlock - 1) / kBlock);
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, gaction.scalar_type(), "tanh_normal_bwd", [&] {
    using T = scalar_t;
    hipLaunchKernelGGL(tanh_normal_bwd_kernel<T>, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                       (const T*)gaction.data_ptr(), glogp.data_ptr<float>(), mean.data_ptr<float>(),
                       logstd.data_ptr<float>(), eps.data_ptr<float>(), scale.data_ptr<float>(),
                       dmean.data_ptr<float>(), dlogstd.data_ptr<float>(), B, A, (float)lmin,
                       (float)lmax);
  });
  return {dmean, dlogstd};
}

// ---------------------------------------------------------------------------
// Fused PPO losses (parity: sheeprl/algos/ppo/loss.py
// — policy_loss :6, value_loss :45, entropy_loss :65).  One reduction kernel
// computes the clipped policy loss, (optionally clipped) value loss and
// entropy loss plus their weighted total; backward recomputes the branch
// selections elementwise.  Gradient semantics match the torch composition
// exactly, including torch.maximum's 0.5/0.5 split on ties — the UNCLIPPED
// region makes the two policy branches equal, so ties are the common case,
// not the edge case.
__global__ void ppo_loss_fwd_kernel(const float* __restrict__ lp_new, const float* __restrict__ lp_old,
                                    const float* __restrict__ adv, const float* __restrict__ v_new,
                                    const float* __restrict__ v_old, const float* __restrict__ ret,
                                    const float* __restrict__ ent, float* __restrict__ out, long N,
                                    float clip, bool clip_vloss, float scale) {
  __shared__ float lds[32];
  float pg = 0.f, vl = 0.f, el = 0.f;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < N;
       i += (long)gridDim.x * blockDim.x) {
    const float r = __expf(lp_new[i] - lp_old[i]);
    const float a = adv[i];
    const float rc = fminf(fmaxf(r, 1.f - clip), 1.f + clip);
    pg += fmaxf(-a * r, -a * rc);
    const float dv = v_new[i] - ret[i];
    if (clip_vloss) {
      const float vc = v_old[i] + fminf(fmaxf(v_new[i] - v_old[i], -clip), clip) - ret[i];
      vl += 0.5f * fmaxf(dv * dv, vc * vc);
    } else {
      vl += dv * dv;
    }
    el += -ent[i];
  }
  pg = wave_sum(pg);
  vl = wave_sum(vl);
  el = wave_sum(el);
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  if (lane == 0) {
    lds[wid] = pg;
    lds[8 + wid] = vl;
    lds[16 + wid] = el;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float p = 0.f, v = 0.f, e = 0.f;
    for (int w = 0; w < (int)(blockDim.x >> 6); ++w) {
      p += lds[w];
      v += lds[8 + w];
      e += lds[16 + w];
    }
    atomicAdd(&out[0], p * scale);
    atomicAdd(&out[1], v * scale);
    atomicAdd(&out[2], e * scale);
  }
}

__global__ void ppo_loss_bwd_kernel(const float* __restrict__ g3, const float* __restrict__ lp_new,
                                    const float* __restrict__ lp_old, const float* __restrict__ adv,
                                    const float* __restrict__ v_new, const float* __restrict__ v_old,
                                    const float* __restrict__ ret, float* __restrict__ dlp,
                                    float* __restrict__ dv_out, float* __restrict__ dent, long N,
                                    float clip, bool clip_vloss, float scale) {
  // g3 = {g_pg, g_v, g_ent} already weighted by the caller's combine coefs
  const float gp = g3[0] * scale, gv = g3[1] * scale, ge = g3[2] * scale;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < N;
       i += (long)gridDim.x * blockDim.x) {
    const float r = __expf(lp_new[i] - lp_old[i]);
    const float a = adv[i];
    const float rc = fminf(fmaxf(r, 1.f - clip), 1.f + clip);
    const float inside = (r >= 1.f - clip && r <= 1.f + clip) ? 1.f : 0.f;
    const float l1 = -a * r, l2 = -a * rc;
    float dr;
    if (l1 > l2) {
      dr = -a;
    } else if (l2 > l1) {
      dr = -a * inside;
    } else {
      dr = -a * 0.5f * (1.f + inside);
    }
    dlp[i] = gp * dr * r;  // dr/dlp_new = r
    const float d = v_new[i] - ret[i];
    if (clip_vloss) {
      const float delta = v_new[i] - v_old[i];
      const float din = (delta >= -clip && delta <= clip) ? 1.f : 0.f;
      const float vc = v_old[i] + fminf(fmaxf(delta, -clip), clip) - ret[i];
      const float u = d * d, c2 = vc * vc;
      float g;
      if (u > c2) {
        g = 2.f * d;
      } else if (c2 > u) {
        g = 2.f * vc * din;
      } else {
        g = 0.5f * (2.f * d + 2.f * vc * din);
      }
      dv_out[i] = gv * 0.5f * g;
    } else {
      dv_out[i] = gv * 2.f * d;
    }
    dent[i] = -ge;
  }
}

std::vector<torch::Tensor> ppo_loss_fwd(const torch::Tensor& lp_new, const torch::Tensor& lp_old,
                                        const torch::Tensor& adv, const torch::Tensor& v_new,
                                        const torch::Tensor& v_old, const torch::Tensor& ret,
                                        const torch::Tensor& ent, double clip, bool clip_vloss,
                                        bool mean) {
  CHECK_IN(lp_new);
  const long N = lp_new.numel();
  TORCH_CHECK(lp_old.numel() == N && adv.numel() == N && v_new.numel() == N && v_old.numel() == N &&
              ret.numel() == N && ent.numel() == N);
  auto out = torch::zeros({3}, lp_new.options());
  const float scale = mean ? 1.f / (float)N : 1.f;
  const int blocks = (int)std::min<long>((N + kBlock - 1) / kBlock, 2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(ppo_loss_fwd_kernel, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                     lp_new.data_ptr<float>(), lp_old.data_ptr<float>(), adv.data_ptr<float>(),
                     v_new.data_ptr<float>(), v_old.data_ptr<float>(), ret.data_ptr<float>(),
                     ent.data_ptr<float>(), out.data_ptr<float>(), N, (float)clip, clip_vloss, scale);
  return {out};
}

std::vector<torch::Tensor> ppo_loss_bwd(const torch::Tensor& g3, const torch::Tensor& lp_new,
                                        const torch::Tensor& lp_old, const torch::Tensor& adv,
                                        const torch::Tensor& v_new, const torch::Tensor& v_old,
                                        const torch::Tensor& ret, double clip, bool clip_vloss,
                                        bool mean) {
  const long N = lp_new.numel();
  auto dlp = torch::empty_like(lp_new);
  auto dv = torch::empty_like(v_new);
  auto dent = torch::empty_like(lp_new);
  const float scale = mean ? 1.f / (float)N : 1.f;
  const int blocks = (int)std::min<long>((N + kBlock - 1) / kBlock, 2048);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(ppo_loss_bwd_kernel, dim3(blocks), dim3(kBlock), 0, stream.stream(),
                     g3.data_ptr<float>(), lp_new.data_ptr<float>(), lp_old.data_ptr<float>(),
                     adv.data_ptr<float>(), v_new.data_ptr<float>(), v_old.data_ptr<float>(),
                     ret.data_ptr<float>(), dlp.data_ptr<float>(), dv.data_ptr<float>(),
                     dent.data_ptr<float>(), N, (float)clip, clip_vloss, scale);
  return {dlp, dv, dent};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("symlog_fwd", [](const torch::Tensor& x) { return symmath<0>(x, c10::nullopt); });
  m.def("symlog_bwd", [](const torch::Tensor& x, const torch::Tensor& g) { return symmath<1>(x, g); });
  m.def("symexp_fwd", [](const torch::Tensor& x) { return symmath<2>(x, c10::nullopt); });
  m.def("symexp_bwd", [](const torch::Tensor& x, const torch::Tensor& g) { return symmath<3>(x, g); });
  m.def("ln_act_fwd", &ln_act_fwd);
  m.def("ln_act_bwd", &ln_act_bwd);
  m.def("gru_gates_fwd", &gru_gates_fwd);
  m.def("gru_gates_bwd", &gru_gates_bwd);
  m.def("gae_scan", &gae_scan);
  m.def("lambda_scan_fwd", &lambda_scan_fwd);
  m.def("lambda_scan_bwd", &lambda_scan_bwd);
  m.def("adam_step", &adam_step);
  m.def("adam_step_dev", &adam_step_dev);
  m.def("adam_step_mt", &adam_step_mt);
  m.def("rmsprop_step_mt", &rmsprop_step_mt);
  m.def("clip_grad_norm_mt", &clip_grad_norm_mt);
  m.def("cat_st_resets_fwd", &cat_st_resets_fwd);
  m.def("cat_st_resets_bwd", &cat_st_resets_bwd);
  m.def("cat_st_fwd", &cat_st_fwd);
  m.def("cat_st_bwd", &cat_st_bwd);
  m.def("masked_lerp_fwd", &masked_lerp_fwd);
  m.def("masked_lerp_bwd", &masked_lerp_bwd);
  // fused-scan variants (strided views / caller-provided outputs+accumulators)
  m.def("ln_act_fwd_o", &ln_act_fwd_o);
  m.def("ln_act_bwd_acc", &ln_act_bwd_acc);
  m.def("gru_gates_fwd_o", &gru_gates_fwd_o, py::arg("y"), py::arg("h"), py::arg("w"), py::arg("b"),
        py::arg("eps"), py::arg("hout"), py::arg("hout2"), py::arg("mean"), py::arg("rstd"),
        py::arg("hout3") = c10::nullopt);
  m.def("gru_gates_bwd_acc", &gru_gates_bwd_acc);
  m.def("cat_st_fwd_o", &cat_st_fwd_o, py::arg("raw"), py::arg("urand"), py::arg("unimix"),
        py::arg("m"), py::arg("onehot"), py::arg("s"), py::arg("onehot2") = c10::nullopt);
  m.def("cat_st_bwd_o", &cat_st_bwd_o);
  m.def("ema_update", &ema_update);
  m.def("obs_norm", &obs_norm);
  m.def("chlast_bias_sum", &chlast_bias_sum);
  m.def("pk_gemm16_test", &pk_gemm16_test);
  m.def("pk_barrier_test", &pk_barrier_test);
  m.def("pk_scan_fwd", &pk_scan_fwd);
  m.def("scan_resets_fwd", &scan_resets_fwd);
  m.def("scan_resets_bwd", &scan_resets_bwd);
  m.def("twohot_lp_fwd", &twohot_lp_fwd);
  m.def("twohot_lp_bwd", &twohot_lp_bwd);
  m.def("klbal_fwd", &klbal_fwd);
  m.def("klbal_bwd", &klbal_bwd);
  m.def("transpose2d", &transpose2d);
  m.def("nll_fwd", &nll_fwd);
  m.def("nll_bwd", &nll_bwd);
  m.def("twohot_mean", &twohot_mean);
  m.def("reinforce_fwd", &reinforce_fwd);
  m.def("reinforce_bwd", &reinforce_bwd);
  m.def("moments_update", &moments_update);
  m.def("tanh_normal_fwd", &tanh_normal_fwd);
  m.def("tanh_normal_bwd", &tanh_normal_bwd);
  m.def("ppo_loss_fwd", &ppo_loss_fwd);
  m.def("ppo_loss_bwd", &ppo_loss_bwd);
  m.def("vloss2_fwd", &vloss2_fwd);
  m.def("vloss2_bwd", &vloss2_bwd);
  m.def("replay_gather", &replay_gather);
  m.def("lstm_gates_fwd", &lstm_gates_fwd);
  m.def("lstm_gates_bwd", &lstm_gates_bwd);
  m.def("scan3_f1", &scan3_f1);
  m.def("scan3_f2", &scan3_f2);
  m.def("scan3_f3", &scan3_f3);
  m.def("scan3_f4", &scan3_f4);
  m.def("scan3_b4", &scan3_b4);
  m.def("scan2_f1", &scan2_f1);
  m.def("scan2_f2", &scan2_f2);
  m.def("scan2_f3", &scan2_f3);
  m.def("scan2_f4", &scan2_f4);
  m.def("scan2_b4", &scan2_b4);
  m.def("scan2_b3", &scan2_b3);
  m.def("scan2_b2", &scan2_b2);
  m.def("scan2_b1", &scan2_b1);
  m.def("g16_plain", &g16_plain);
  m.def("g16_splitk", &g16_splitk);
  m.def("g16_ln_silu", &g16_ln_silu);
  m.def("g16_cat_st", &g16_cat_st);
}
