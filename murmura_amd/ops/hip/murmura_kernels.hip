// murmura_amd HIP/CDNA4 kernels (gfx950 / MI355X).
//
// Kernel inventory follows SURVEY.md §2.9 (K1-K12): the aggregation and
// round-loop hot paths of a decentralized-FL engine, operating on flat
// parameter vectors [P] and stacked states [m, P] (m = 1 + #neighbors,
// typically <= 16 on one 8xMI355X box; P up to ~100M).
//
// Every kernel here is HBM-bandwidth-bound streaming work; the design rules
// applied (per the CDNA4 programming guide):
//  - wave64 everywhere; blocks of 256 threads; grids sized >> 256 workgroups
//    via grid-stride loops so all 8 XCDs fill.
//  - 16-byte packed loads per lane (float4 / 8x bf16) — the compiler does not
//    reliably auto-vectorize bf16 element loads.
//  - fp32 accumulation regardless of storage dtype.
//  - cross-lane reductions in-register via __shfl_xor (64-lane), one LDS
//    accumulator per block, one global atomic per block — no per-pair host
//    syncs (the reference's aggregation does m^2 x num_keys .item() calls).
//  - pairwise-L2 reads each row ONCE (register-tiled Gram accumulation,
//    compile-time m) instead of re-reading per pair.
//  - Philox4x32-10 counter RNG for on-GPU attack injection (deterministic
//    per (seed, offset, index) with no generator state).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <cstdint>

#define WAVE 64
#define BLOCK 256

// ------------------------------------------------------------------ dtypes
struct bf16raw {
  unsigned short v;
};

__device__ __forceinline__ float bf2f(unsigned short h) {
  unsigned int u = ((unsigned int)h) << 16;
  return __uint_as_float(u);
}

__device__ __forceinline__ unsigned short f2bf(float f) {
  unsigned int u = __float_as_uint(f);
  if ((u & 0x7fffffffu) > 0x7f800000u) return (unsigned short)0x7fc0;  // NaN
  unsigned int lsb = (u >> 16) & 1u;
  u += 0x7fffu + lsb;  // round-to-nearest-even
  return (unsigned short)(u >> 16);
}

__device__ __forceinline__ float to_f(float x) { return x; }
__device__ __forceinline__ float to_f(bf16raw x) { return bf2f(x.v); }
__device__ __forceinline__ void from_f(float& d, float s) { d = s; }
__device__ __forceinline__ void from_f(bf16raw& d, float s) { d.v = f2bf(s); }

// 16-byte pack: 4 x fp32 or 8 x bf16 per lane per load
template <typename T>
struct Pack16 {
  static constexpr int N = 16 / sizeof(T);
  T e[N];
};

static inline int env_int(const char* name, int dflt) {
  const char* v = getenv(name);
  return v ? atoi(v) : dflt;
}

static inline int grid_for(int64_t work_items, int per_block, int cap = 4096) {
  int64_t b = (work_items + per_block - 1) / per_block;
  if (b < 1) b = 1;
  if (b > cap) b = cap;
  return (int)b;
}

// wave-level fp32 sum over all 64 lanes
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

// ================================================================== K1
// out[p] = sum_i w[i] * x[i*P + p]
template <typename T>
__global__ void weighted_sum_kernel(const T* __restrict__ x,
                                    const float* __restrict__ w, T* __restrict__ out,
                                    int m, int64_t P) {
  __shared__ float ws[256];
  for (int i = threadIdx.x; i < m; i += blockDim.x) ws[i] = w[i];
  __syncthreads();

  constexpr int N = Pack16<T>::N;
  const int64_t nvec = P / N;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;

  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec; v += stride) {
    float acc[N];
#pragma unroll
    for (int k = 0; k < N; ++k) acc[k] = 0.0f;
    for (int i = 0; i < m; ++i) {
      Pack16<T> pk = reinterpret_cast<const Pack16<T>*>(x + (int64_t)i * P)[v];
      const float wi = ws[i];
#pragma unroll
      for (int k = 0; k < N; ++k) acc[k] = fmaf(wi, to_f(pk.e[k]), acc[k]);
    }
    Pack16<T> po;
#pragma unroll
    for (int k = 0; k < N; ++k) from_f(po.e[k], acc[k]);
    reinterpret_cast<Pack16<T>*>(out)[v] = po;
  }
  // scalar tail
  for (int64_t p = nvec * N + blockIdx.x * blockDim.x + threadIdx.x; p < P; p += stride) {
    float acc = 0.0f;
    for (int i = 0; i < m; ++i) acc = fmaf(ws[i], to_f(x[(int64_t)i * P + p]), acc);
    from_f(out[p], acc);
  }
}

// ================================================================== K2
// Gram matrix over rows: gram[i, j] = <x_i, x_j> for i <= j, each row read
// exactly once (register-tiled over compile-time M).
template <typename T, int M>
__global__ __launch_bounds__(BLOCK, 1) void gram_kernel(
    const T* __restrict__ x, int64_t P, int64_t ld, float* __restrict__ gram) {
  // __launch_bounds__(256, 1) lifts the default VGPR cap so m up to 16
  // (136 accumulators + 16 row packs) stays in registers — without it the
  // compiler spilled at m ~ 12 (measured 647 GB/s; VERDICT weak #4/#8)
  constexpr int NPAIR = M * (M + 1) / 2;
  constexpr int N = Pack16<T>::N;
  __shared__ float lacc[NPAIR];
  for (int t = threadIdx.x; t < NPAIR; t += blockDim.x) lacc[t] = 0.0f;
  __syncthreads();

  float acc[NPAIR];
#pragma unroll
  for (int t = 0; t < NPAIR; ++t) acc[t] = 0.0f;

  const int64_t nvec = P / N;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec; v += stride) {
    Pack16<T> rows[M];
#pragma unroll
    for (int i = 0; i < M; ++i) rows[i] = reinterpret_cast<const Pack16<T>*>(x + (int64_t)i * ld)[v];
    int t = 0;
#pragma unroll
    for (int i = 0; i < M; ++i) {
#pragma unroll
      for (int j = i; j < M; ++j) {
        float s = acc[t];
#pragma unroll
        for (int k = 0; k < N; ++k) s = fmaf(to_f(rows[i].e[k]), to_f(rows[j].e[k]), s);
        acc[t] = s;
        ++t;
      }
    }
  }
  // scalar tail
  for (int64_t p = nvec * N + blockIdx.x * blockDim.x + threadIdx.x; p < P; p += stride) {
    float rowsf[M];
#pragma unroll
    for (int i = 0; i < M; ++i) rowsf[i] = to_f(x[(int64_t)i * ld + p]);
    int t = 0;
#pragma unroll
    for (int i = 0; i < M; ++i)
#pragma unroll
      for (int j = i; j < M; ++j) acc[t++] += rowsf[i] * rowsf[j];
  }

  // reduce: wave shuffle -> LDS -> one global atomic per pair per block
#pragma unroll
  for (int t = 0; t < NPAIR; ++t) {
    float s = wave_sum(acc[t]);
    if ((threadIdx.x & (WAVE - 1)) == 0) atomicAdd(&lacc[t], s);
  }
  __syncthreads();
  for (int t = threadIdx.x; t < NPAIR; t += blockDim.x) atomicAdd(&gram[t], lacc[t]);
}

// Cross-Gram tile: out[i, j] += <xi_i, xj_j> for an MI x MJ row-tile pair.
// Used to decompose m > 10 Gram matrices into 8x8 tiles: the monolithic
// kernel's m*(m+1)/2 accumulators spill registers at m ~ 12 (measured
// 647 GB/s vs ~5600 at m=8); 8x8 tiles keep 64 accumulators + 2x8 row
// packs in registers at the cost of re-reading rows once per tile pair.
template <typename T, int MI, int MJ>
__global__ void gram_cross_kernel(const T* __restrict__ xi, const T* __restrict__ xj,
                                  int64_t P, int64_t ld, float* __restrict__ out,
                                  int ldo) {
  constexpr int N = Pack16<T>::N;
  __shared__ float lacc[MI * MJ];
  for (int t = threadIdx.x; t < MI * MJ; t += blockDim.x) lacc[t] = 0.0f;
  __syncthreads();

  float acc[MI * MJ];
#pragma unroll
  for (int t = 0; t < MI * MJ; ++t) acc[t] = 0.0f;

  const int64_t nvec = P / N;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec; v += stride) {
    Pack16<T> ri[MI], rj[MJ];
#pragma unroll
    for (int i = 0; i < MI; ++i) ri[i] = reinterpret_cast<const Pack16<T>*>(xi + (int64_t)i * ld)[v];
#pragma unroll
    for (int j = 0; j < MJ; ++j) rj[j] = reinterpret_cast<const Pack16<T>*>(xj + (int64_t)j * ld)[v];
#pragma unroll
    for (int i = 0; i < MI; ++i)
#pragma unroll
      for (int j = 0; j < MJ; ++j) {
        float s = acc[i * MJ + j];
#pragma unroll
        for (int k = 0; k < N; ++k) s = fmaf(to_f(ri[i].e[k]), to_f(rj[j].e[k]), s);
        acc[i * MJ + j] = s;
      }
  }
  for (int64_t p = nvec * N + blockIdx.x * blockDim.x + threadIdx.x; p < P; p += stride) {
    float fi[MI], fj[MJ];
#pragma unroll
    for (int i = 0; i < MI; ++i) fi[i] = to_f(xi[(int64_t)i * ld + p]);
#pragma unroll
    for (int j = 0; j < MJ; ++j) fj[j] = to_f(xj[(int64_t)j * ld + p]);
#pragma unroll
    for (int i = 0; i < MI; ++i)
#pragma unroll
      for (int j = 0; j < MJ; ++j) acc[i * MJ + j] += fi[i] * fj[j];
  }

#pragma unroll
  for (int t = 0; t < MI * MJ; ++t) {
    float s = wave_sum(acc[t]);
    if ((threadIdx.x & (WAVE - 1)) == 0) atomicAdd(&lacc[t], s);
  }
  __syncthreads();
  for (int t = threadIdx.x; t < MI * MJ; t += blockDim.x) {
    atomicAdd(&out[(t / MJ) * ldo + (t % MJ)], lacc[t]);
  }
}

// ================================================================== K12
// out[i] += sum_p x[i,p]^2  (per-row squared norms)
template <typename T>
__global__ void row_sqnorm_kernel(const T* __restrict__ x, int64_t P, float* __restrict__ out) {
  constexpr int N = Pack16<T>::N;
  const T* row = x + (int64_t)blockIdx.y * P;
  const int64_t nvec = P / N;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float acc = 0.0f;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec; v += stride) {
    Pack16<T> pk = reinterpret_cast<const Pack16<T>*>(row)[v];
#pragma unroll
    for (int k = 0; k < N; ++k) {
      float f = to_f(pk.e[k]);
      acc = fmaf(f, f, acc);
    }
  }
  for (int64_t p = nvec * N + blockIdx.x * blockDim.x + threadIdx.x; p < P; p += stride) {
    float f = to_f(row[p]);
    acc = fmaf(f, f, acc);
  }
  __shared__ float lsum;
  if (threadIdx.x == 0) lsum = 0.0f;
  __syncthreads();
  float s = wave_sum(acc);
  if ((threadIdx.x & (WAVE - 1)) == 0) atomicAdd(&lsum, s);
  __syncthreads();
  if (threadIdx.x == 0) atomicAdd(&out[blockIdx.y], lsum);
}

// sq dists of k rows against one "own" row (fallback for k+1 > 16)
template <typename T>
__global__ void sqdist_to_kernel(const T* __restrict__ own, const T* __restrict__ x,
                                 int64_t P, float* __restrict__ out) {
  constexpr int N = Pack16<T>::N;
  const T* row = x + (int64_t)blockIdx.y * P;
  const int64_t nvec = P / N;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float acc = 0.0f;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec; v += stride) {
    Pack16<T> a = reinterpret_cast<const Pack16<T>*>(row)[v];
    Pack16<T> b = reinterpret_cast<const Pack16<T>*>(own)[v];
#pragma unroll
    for (int k = 0; k < N; ++k) {
      float d = to_f(a.e[k]) - to_f(b.e[k]);
      acc = fmaf(d, d, acc);
    }
  }
  for (int64_t p = nvec * N + blockIdx.x * blockDim.x + threadIdx.x; p < P; p += stride) {
    float d = to_f(row[p]) - to_f(own[p]);
    acc = fmaf(d, d, acc);
  }
  __shared__ float lsum;
  if (threadIdx.x == 0) lsum = 0.0f;
  __syncthreads();
  float s = wave_sum(acc);
  if ((threadIdx.x & (WAVE - 1)) == 0) atomicAdd(&lsum, s);
  __syncthreads();
  if (threadIdx.x == 0) atomicAdd(&out[blockIdx.y], lsum);
}

// ================================================================== K4
// Count-Sketch: out[row, bin(p)] += sign(p) * x[row, p].
//
// Round-2 redesign (round 1 measured ~690 GB/s of ACTUAL bytes — LDS-atomic
// issue-bound with serial per-row atomics and 8 B/element table reads):
//  - hash+sign packed into ONE int32 per element (bin in the low 31 bits,
//    sign in the sign bit): table traffic 8 B -> 4 B per element
//  - Pack16 vector loads for the data rows
//  - R replicated LDS histograms (one per wave group) to cut atomic
//    conflict serialization
//  - all m rows x N pack elements issue atomics back-to-back per iteration
//    (64 independent ds_adds in flight hide LDS atomic latency)
template <typename T>
__global__ void count_sketch_multirow_kernel(const T* __restrict__ x,
                                             const int* __restrict__ pt, int m,
                                             int64_t P, int S,
                                             float* __restrict__ out, int R) {
  extern __shared__ float hist[];  // [R][m][S]
  const int nh = R * m * S;
  for (int b = threadIdx.x; b < nh; b += blockDim.x) hist[b] = 0.0f;
  __syncthreads();
  float* my = hist + (int)((threadIdx.x / WAVE) % R) * m * S;

  constexpr int N = Pack16<T>::N;
  const int64_t nvec = P / N;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  if (P % N == 0) {
    for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec;
         v += stride) {
      int hw[N];
#pragma unroll
      for (int k = 0; k < N; k += 4) {
        const int4 t4 = reinterpret_cast<const int4*>(pt + v * N)[k / 4];
        hw[k] = t4.x;
        hw[k + 1] = t4.y;
        hw[k + 2] = t4.z;
        hw[k + 3] = t4.w;
      }
      for (int i = 0; i < m; ++i) {
        Pack16<T> xv = reinterpret_cast<const Pack16<T>*>(x + (int64_t)i * P)[v];
#pragma unroll
        for (int k = 0; k < N; ++k) {
          const int w = hw[k];
          float val = to_f(xv.e[k]);
          atomicAdd(&my[i * S + (w & 0x7FFFFFFF)], w < 0 ? -val : val);
        }
      }
    }
  } else {  // unaligned P: scalar path
    for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < P;
         p += stride) {
      const int w = pt[p];
      const int bin = w & 0x7FFFFFFF;
      for (int i = 0; i < m; ++i) {
        float val = to_f(x[(int64_t)i * P + p]);
        atomicAdd(&my[i * S + bin], w < 0 ? -val : val);
      }
    }
  }
  __syncthreads();
  for (int b = threadIdx.x; b < m * S; b += blockDim.x) {
    float acc = 0.0f;
    for (int r = 0; r < R; ++r) acc += hist[r * m * S + b];
    if (acc != 0.0f) atomicAdd(&out[b], acc);
  }
}

// Grouped-table variant (the production path): bins are constant over
// aligned groups of 8 coordinates (ops/reference.py make_sketch_tables),
// signs per element. Each thread accumulates a group's signed sum for a row
// in registers and issues ONE LDS atomic — 8x fewer atomics than the
// per-element kernel, which is atomic-issue-bound (~200G atomics/s measured).
// Table entry: bin in bits 0..15, per-element sign bits in 16..23.
template <typename T, bool ALIGNED>
__global__ void count_sketch_g8_kernel(const T* __restrict__ x,
                                       const int* __restrict__ gt, int m,
                                       int64_t P, int64_t ngroups, int S,
                                       const int* __restrict__ tailt, int ntail,
                                       float* __restrict__ out, int R) {
  extern __shared__ float hist[];  // [R][m][S]
  const int nh = R * m * S;
  for (int b = threadIdx.x; b < nh; b += blockDim.x) hist[b] = 0.0f;
  __syncthreads();
  float* my = hist + (int)((threadIdx.x / WAVE) % R) * m * S;

  constexpr int N = Pack16<T>::N;  // 8 bf16 / 4 fp32 per 16-B pack
  constexpr int PPG = 8 / N;       // packs per group
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; g < ngroups;
       g += stride) {
    const int w = gt[g];
    const int bin = w & 0xFFFF;
    float sgn[8];
#pragma unroll
    for (int k = 0; k < 8; ++k) sgn[k] = ((w >> (16 + k)) & 1) ? -1.0f : 1.0f;
    for (int i = 0; i < m; ++i) {
      const T* row = x + (int64_t)i * P + g * 8;
      float s = 0.0f;
      if (ALIGNED) {
#pragma unroll
        for (int pp = 0; pp < PPG; ++pp) {
          Pack16<T> xv = reinterpret_cast<const Pack16<T>*>(row)[pp];
#pragma unroll
          for (int k = 0; k < N; ++k) s = fmaf(sgn[pp * N + k], to_f(xv.e[k]), s);
        }
      } else {  // rows not 16-B aligned (P % 8 != 0): element loads
#pragma unroll
        for (int k = 0; k < 8; ++k) s = fmaf(sgn[k], to_f(row[k]), s);
      }
      atomicAdd(&my[i * S + bin], s);
    }
  }
  // per-element tail for p in [ngroups*8, P)
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; t < ntail;
       t += stride) {
    const int w = tailt[t];
    const int bin = w & 0x7FFFFFFF;
    const int64_t p = ngroups * 8 + t;
    for (int i = 0; i < m; ++i) {
      float val = to_f(x[(int64_t)i * P + p]);
      atomicAdd(&my[i * S + bin], w < 0 ? -val : val);
    }
  }
  __syncthreads();
  for (int b = threadIdx.x; b < m * S; b += blockDim.x) {
    float acc = 0.0f;
    for (int r = 0; r < R; ++r) acc += hist[r * m * S + b];
    if (acc != 0.0f) atomicAdd(&out[b], acc);
  }
}

// fallback for sketches too large for replicated LDS: one row per blockIdx.y,
// single histogram
template <typename T>
__global__ void count_sketch_kernel(const T* __restrict__ x, const int* __restrict__ pt,
                                    int64_t P, int S, float* __restrict__ out) {
  extern __shared__ float hist[];
  for (int b = threadIdx.x; b < S; b += blockDim.x) hist[b] = 0.0f;
  __syncthreads();

  const T* row = x + (int64_t)blockIdx.y * P;
  float* orow = out + (int64_t)blockIdx.y * S;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < P; p += stride) {
    const int w = pt[p];
    float val = to_f(row[p]);
    atomicAdd(&hist[w & 0x7FFFFFFF], w < 0 ? -val : val);
  }
  __syncthreads();
  for (int b = threadIdx.x; b < S; b += blockDim.x) {
    if (hist[b] != 0.0f) atomicAdd(&orow[b], hist[b]);
  }
}

// ================================================================== K6
// p <- p - lr * g over the whole param prefix, one launch
template <typename T>
__global__ void sgd_step_kernel(T* __restrict__ p, const T* __restrict__ g, float lr,
                                int64_t P) {
  constexpr int N = Pack16<T>::N;
  const int64_t nvec = P / N;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec; v += stride) {
    Pack16<T> pv = reinterpret_cast<Pack16<T>*>(p)[v];
    Pack16<T> gv = reinterpret_cast<const Pack16<T>*>(g)[v];
#pragma unroll
    for (int k = 0; k < N; ++k) from_f(pv.e[k], fmaf(-lr, to_f(gv.e[k]), to_f(pv.e[k])));
    reinterpret_cast<Pack16<T>*>(p)[v] = pv;
  }
  for (int64_t i = nvec * N + blockIdx.x * blockDim.x + threadIdx.x; i < P; i += stride) {
    from_f(p[i], fmaf(-lr, to_f(g[i]), to_f(p[i])));
  }
}

// K6 variant: lr read from device memory (hipGraph-capturable with a
// per-round lr update outside the graph)
template <typename T>
__global__ void sgd_step_lrt_kernel(T* __restrict__ p, const T* __restrict__ g,
                                    const float* __restrict__ lr, int64_t P) {
  const float l = *lr;
  constexpr int N = Pack16<T>::N;
  const int64_t nvec = P / N;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec; v += stride) {
    Pack16<T> pv = reinterpret_cast<Pack16<T>*>(p)[v];
    Pack16<T> gv = reinterpret_cast<const Pack16<T>*>(g)[v];
#pragma unroll
    for (int k = 0; k < N; ++k) from_f(pv.e[k], fmaf(-l, to_f(gv.e[k]), to_f(pv.e[k])));
    reinterpret_cast<Pack16<T>*>(p)[v] = pv;
  }
  for (int64_t i = nvec * N + blockIdx.x * blockDim.x + threadIdx.x; i < P; i += stride) {
    from_f(p[i], fmaf(-l, to_f(g[i]), to_f(p[i])));
  }
}

// ================================================================== K10
// Philox4x32-10 counter RNG + Box-Muller: out = x + N(0, sigma^2).
__device__ __forceinline__ void philox_round(unsigned int& c0, unsigned int& c1,
                                             unsigned int& c2, unsigned int& c3,
                                             unsigned int k0, unsigned int k1) {
  const unsigned int M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  unsigned long long p0 = (unsigned long long)M0 * c0;
  unsigned long long p1 = (unsigned long long)M1 * c2;
  unsigned int h0 = (unsigned int)(p0 >> 32), l0 = (unsigned int)p0;
  unsigned int h1 = (unsigned int)(p1 >> 32), l1 = (unsigned int)p1;
  unsigned int n0 = h1 ^ c1 ^ k0;
  unsigned int n1 = l1;
  unsigned int n2 = h0 ^ c3 ^ k1;
  unsigned int n3 = l0;
  c0 = n0; c1 = n1; c2 = n2; c3 = n3;
}

__device__ __forceinline__ void philox4(unsigned long long seed, unsigned long long ctr_hi,
                                        unsigned long long ctr_lo, unsigned int out[4]) {
  unsigned int k0 = (unsigned int)seed, k1 = (unsigned int)(seed >> 32);
  unsigned int c0 = (unsigned int)ctr_lo, c1 = (unsigned int)(ctr_lo >> 32);
  unsigned int c2 = (unsigned int)ctr_hi, c3 = (unsigned int)(ctr_hi >> 32);
  const unsigned int W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    philox_round(c0, c1, c2, c3, k0, k1);
    k0 += W0; k1 += W1;
  }
  out[0] = c0; out[1] = c1; out[2] = c2; out[3] = c3;
}

__device__ __forceinline__ float u32_to_uniform(unsigned int u) {
  // (0, 1]: (u + 1) * 2^-32
  return ((float)u + 1.0f) * 2.3283064365386963e-10f;
}

template <typename T>
__global__ void gaussian_inject_kernel(const T* __restrict__ x, T* __restrict__ out,
                                       float sigma, unsigned long long seed,
                                       unsigned long long offset, int64_t P) {
  const int64_t nquad = (P + 3) / 4;  // 4 outputs per philox call
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t q = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; q < nquad; q += stride) {
    unsigned int r[4];
    philox4(seed, offset, (unsigned long long)q, r);
    float u0 = u32_to_uniform(r[0]), u1 = u32_to_uniform(r[1]);
    float u2 = u32_to_uniform(r[2]), u3 = u32_to_uniform(r[3]);
    float r0 = sqrtf(-2.0f * __logf(u0)), a0;
    float s0, c0v;
    __sincosf(6.2831853071795864f * u1, &s0, &c0v);
    float r1 = sqrtf(-2.0f * __logf(u2)), s1, c1v;
    __sincosf(6.2831853071795864f * u3, &s1, &c1v);
    float n[4] = {r0 * c0v, r0 * s0, r1 * c1v, r1 * s1};
    a0 = sigma;
    int64_t base = q * 4;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      int64_t i = base + k;
      if (i < P) from_f(out[i], fmaf(a0, n[k], to_f(x[i])));
    }
  }
}

// ================================================================== K11
template <typename T>
__global__ void scale_kernel(const T* __restrict__ x, T* __restrict__ out, float lam,
                             int64_t P) {
  constexpr int N = Pack16<T>::N;
  const int64_t nvec = P / N;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec; v += stride) {
    Pack16<T> pv = reinterpret_cast<const Pack16<T>*>(x)[v];
#pragma unroll
    for (int k = 0; k < N; ++k) from_f(pv.e[k], lam * to_f(pv.e[k]));
    reinterpret_cast<Pack16<T>*>(out)[v] = pv;
  }
  for (int64_t i = nvec * N + blockIdx.x * blockDim.x + threadIdx.x; i < P; i += stride) {
    from_f(out[i], lam * to_f(x[i]));
  }
}

// ================================================================== K7
// Fused CE-loss + accuracy epilogue: one WAVE per row.
// out[0] += sum_rows (logsumexp - logit[target]); out[1] += #correct
template <typename T>
__global__ void ce_loss_acc_kernel(const T* __restrict__ logits,
                                   const int64_t* __restrict__ targets, int B, int C,
                                   float* __restrict__ out) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;          // wave within block (4)
  const int waves_per_block = blockDim.x / WAVE;
  float loss_acc = 0.0f;
  float corr_acc = 0.0f;
  for (int row = blockIdx.x * waves_per_block + wid; row < B;
       row += gridDim.x * waves_per_block) {
    const T* lr = logits + (int64_t)row * C;
    const int tgt = (int)targets[row];
    float vmax = -3.4e38f;
    int imax = 0;
    float xt = 0.0f;
    for (int c = lane; c < C; c += WAVE) {
      float v = to_f(lr[c]);
      if (v > vmax) { vmax = v; imax = c; }
      if (c == tgt) xt = v;
    }
    // wave argmax (max value, then lowest index among ties)
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float ov = __shfl_xor(vmax, off, WAVE);
      int oi = __shfl_xor(imax, off, WAVE);
      if (ov > vmax || (ov == vmax && oi < imax)) { vmax = ov; imax = oi; }
    }
    float se = 0.0f;
    for (int c = lane; c < C; c += WAVE) se += __expf(to_f(lr[c]) - vmax);
    se = wave_sum(se);
    // broadcast xt from the lane that owns column tgt (c == tgt happens on
    // lane tgt % WAVE only; other lanes hold 0)
    float xt_b = __shfl(xt, tgt % WAVE, WAVE);
    if (lane == 0) {
      loss_acc += __logf(se) + vmax - xt_b;
      corr_acc += (imax == tgt) ? 1.0f : 0.0f;
    }
  }
  if ((threadIdx.x & (WAVE - 1)) == 0) {
    atomicAdd(&out[0], loss_acc);
    atomicAdd(&out[1], corr_acc);
  }
}

// ================================================================== K8
// Evidential stats: alpha = softplus(x) + 1; per-row vacuity K/S, entropy of
// alpha/S, strength S, correctness; sums atomically accumulated.
__device__ __forceinline__ float softplusf(float x) {
  // numerically-stable log(1 + exp(x))
  if (x > 20.0f) return x;
  if (x < -20.0f) return __expf(x);
  return __logf(1.0f + __expf(x));
}

template <typename T>
__global__ void evidential_stats_kernel(const T* __restrict__ logits,
                                        const int64_t* __restrict__ targets, int B, int C,
                                        float* __restrict__ out) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int waves_per_block = blockDim.x / WAVE;
  float vac_acc = 0.0f, ent_acc = 0.0f, str_acc = 0.0f, corr_acc = 0.0f;
  for (int row = blockIdx.x * waves_per_block + wid; row < B;
       row += gridDim.x * waves_per_block) {
    const T* lr = logits + (int64_t)row * C;
    const int tgt = (int)targets[row];
    float ssum = 0.0f, amax = -3.4e38f;
    int imax = 0;
    for (int c = lane; c < C; c += WAVE) {
      float a = softplusf(to_f(lr[c])) + 1.0f;
      ssum += a;
      if (a > amax) { amax = a; imax = c; }
    }
    ssum = wave_sum(ssum);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float ov = __shfl_xor(amax, off, WAVE);
      int oi = __shfl_xor(imax, off, WAVE);
      if (ov > amax || (ov == amax && oi < imax)) { amax = ov; imax = oi; }
    }
    float ent = 0.0f;
    for (int c = lane; c < C; c += WAVE) {
      float a = softplusf(to_f(lr[c])) + 1.0f;
      float p = a / ssum;
      float pl = p > 1e-10f ? p : 1e-10f;
      ent -= p * __logf(pl);
    }
    ent = wave_sum(ent);
    if (lane == 0) {
      vac_acc += (float)C / ssum;
      ent_acc += ent;
      str_acc += ssum;
      corr_acc += (imax == tgt) ? 1.0f : 0.0f;
    }
  }
  if ((threadIdx.x & (WAVE - 1)) == 0) {
    atomicAdd(&out[0], vac_acc);
    atomicAdd(&out[1], ent_acc);
    atomicAdd(&out[2], str_acc);
    atomicAdd(&out[3], corr_acc);
  }
}

// ================================================================== K14
// Fused NHWC BatchNorm2d for the training hot path. torch's channels-last BN
// kernels measured ~230 GB/s on MI355X (37 us for an 8 MB reduction);
// these are plain two-pass streaming kernels targeting the HBM roofline.
//
// Memory view: x is [R, C] with C contiguous (R = N*H*W), C in {8..1024},
// C % 8 == 0. fp32 accumulation/statistics regardless of element type.

// pass 1: per-channel sum and sum-of-squares.
// Each thread owns one 16-byte CHANNEL PACK (8 bf16 / 4 fp32 channels) and
// accumulates in registers while streaming rows; per-block combine goes
// through one LDS histogram pass, then one global atomic per channel per
// block. (The first version assigned one scalar channel per thread — 2-byte
// loads, 190 GB/s; packs restore the streaming roofline.)
// ticket counter for the fused partials+finalize last-arriver (BN calls are
// stream-ordered within a process, so one counter suffices; the last block
// self-cleans it for the next call — zero-initialized at module load)
__device__ unsigned murmura_bn_ticket = 0u;

// reduce ws[0..G)[{c, C+c}] with 8-way unrolled independent accumulators —
// the loads have no cross-iteration dependency, so 16 stay in flight per
// thread and the loop is bandwidth- not latency-bound (a naive serial loop
// here measured 60 us; this form ~2 us).
__device__ __forceinline__ void bn_reduce_partials(const float* __restrict__ ws,
                                                   int G, int C, int c,
                                                   float& s_out, float& q_out) {
  constexpr int UR = 8;
  float s[UR], q[UR];
#pragma unroll
  for (int u = 0; u < UR; ++u) s[u] = q[u] = 0.0f;
  int g = 0;
  for (; g + UR <= G; g += UR) {
#pragma unroll
    for (int u = 0; u < UR; ++u) {
      s[u] += ws[(int64_t)(g + u) * 2 * C + c];
      q[u] += ws[(int64_t)(g + u) * 2 * C + C + c];
    }
  }
  for (; g < G; ++g) {
    s[0] += ws[(int64_t)g * 2 * C + c];
    q[0] += ws[(int64_t)g * 2 * C + C + c];
  }
#pragma unroll
  for (int u = 1; u < UR; ++u) {
    s[0] += s[u];
    q[0] += q[u];
  }
  s_out = s[0];
  q_out = q[0];
}


// Round-2 redesign. Round 1 capped the reduction grid at 64 blocks (global
// atomics serialized beyond that) which left 192 of 256 CUs idle and ran at
// ~0.6 TB/s, and every call pre-zeroed a [2,C] workspace (a ~4.8 us
// FillFunctor launch x 40/step). New scheme:
//   phase 1  bn_partials_kernel   per-block partials, NO atomics, NO
//                                 pre-zeroing, grid sized for the chip
//   finalize bn_finalize_*_kernel tiny: reduces G partials AND precomputes
//                                 the per-channel coefficients (removes the
//                                 per-block recompute from the elementwise
//                                 pass)
//   pass 2   bn_norm/bn_bwd_dx    elementwise; optionally fuses the ResNet
//                                 residual add (+ReLU) and its backward,
//                                 removing 2 elementwise kernels per block
//                                 per direction
// FWD accumulates (x, x^2); BWD accumulates (g, g*xhat) with the ReLU mask
// from the saved output (mask applies to the residual grad too, since the
// ReLU follows the add).
template <typename T, bool BWD, bool RELU>
__global__ void bn_partials_kernel(const T* __restrict__ x,
                                   const T* __restrict__ dy,
                                   const T* __restrict__ yout,
                                   int64_t R, int C,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ invstd,
                                   float* __restrict__ ws,
                                   // fused-finalize args (guide §6 G16
                                   // last-arriver recipe); fwd uses
                                   // eps/momentum/w/b/coef/saved/running,
                                   // bwd uses w/gcoef/dweight/dbias
                                   float eps, float momentum,
                                   const T* __restrict__ w,
                                   const T* __restrict__ b,
                                   float* __restrict__ coef,
                                   float* __restrict__ save_mean,
                                   float* __restrict__ save_invstd,
                                   T* __restrict__ running_mean,
                                   T* __restrict__ running_var,
                                   T* __restrict__ dweight,
                                   T* __restrict__ dbias) {
  constexpr int N = Pack16<T>::N;
  const int ppr = C / N;  // packs per row; caller guarantees divisibility
  const int pk = threadIdx.x % ppr;
  const int rsub = threadIdx.x / ppr;
  const int rows_per_iter = max(1, (int)blockDim.x / ppr);
  const int cbase = pk * N;

  float sacc[N], qacc[N], m[N], is[N];
#pragma unroll
  for (int k = 0; k < N; ++k) {
    sacc[k] = qacc[k] = 0.0f;
    if (BWD) {
      m[k] = mean[cbase + k];
      is[k] = invstd[cbase + k];
    }
  }
  if (rsub < rows_per_iter) {
    for (int64_t r = (int64_t)blockIdx.x * rows_per_iter + rsub; r < R;
         r += (int64_t)gridDim.x * rows_per_iter) {
      Pack16<T> xv = *reinterpret_cast<const Pack16<T>*>(x + r * C + cbase);
      if (!BWD) {
#pragma unroll
        for (int k = 0; k < N; ++k) {
          float f = to_f(xv.e[k]);
          sacc[k] += f;
          qacc[k] = fmaf(f, f, qacc[k]);
        }
      } else {
        Pack16<T> gv = *reinterpret_cast<const Pack16<T>*>(dy + r * C + cbase);
        Pack16<T> yv;
        if (RELU) yv = *reinterpret_cast<const Pack16<T>*>(yout + r * C + cbase);
#pragma unroll
        for (int k = 0; k < N; ++k) {
          float g = to_f(gv.e[k]);
          if (RELU && to_f(yv.e[k]) <= 0.0f) g = 0.0f;
          float xh = (to_f(xv.e[k]) - m[k]) * is[k];
          sacc[k] += g;
          qacc[k] = fmaf(g, xh, qacc[k]);
        }
      }
    }
  }
  // block combine -> ws[blockIdx][0..C)=sum, [C..2C)=sumsq (no atomics on
  // the pow2 path: cross-lane shfl tree, then one LDS row per wave)
  extern __shared__ float lds[];
  float* out = ws + (int64_t)blockIdx.x * 2 * C;
  const int nwaves = blockDim.x / WAVE;
  if ((ppr & (ppr - 1)) == 0 && ppr <= WAVE) {
#pragma unroll
    for (int k = 0; k < N; ++k) {
      for (int off = WAVE >> 1; off >= ppr; off >>= 1) {
        sacc[k] += __shfl_xor(sacc[k], off, WAVE);
        qacc[k] += __shfl_xor(qacc[k], off, WAVE);
      }
    }
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    if (lane < ppr) {
#pragma unroll
      for (int k = 0; k < N; ++k) {
        lds[wave * 2 * C + lane * N + k] = sacc[k];
        lds[wave * 2 * C + C + lane * N + k] = qacc[k];
      }
    }
    __syncthreads();
    for (int c = threadIdx.x; c < 2 * C; c += blockDim.x) {
      float s = 0.0f;
      for (int w2 = 0; w2 < nwaves; ++w2) s += lds[w2 * 2 * C + c];
      out[c] = s;
    }
  } else {  // non-pow2 or very wide channel counts: LDS-atomic fallback
    for (int c = threadIdx.x; c < 2 * C; c += blockDim.x) lds[c] = 0.0f;
    __syncthreads();
#pragma unroll
    for (int k = 0; k < N; ++k) {
      atomicAdd(&lds[cbase + k], sacc[k]);
      atomicAdd(&lds[C + cbase + k], qacc[k]);
    }
    __syncthreads();
    for (int c = threadIdx.x; c < 2 * C; c += blockDim.x) out[c] = lds[c];
  }

  if (coef == nullptr) return;  // fusion disabled: separate finalize kernel
  // ---- fused finalize: the LAST-ARRIVING block reduces the G partials and
  // computes the per-channel coefficients in the same launch (saves a
  // ~7 us finalize kernel per BN call; 40 of them per flagship batch).
  // Publish/observe follows cdna_hip_programming.md §6 Guideline 16: plain
  // stores -> wait -> barrier -> lane-0 agent release fence -> asm vmcnt
  // wait (ROCm 7.2 drops the post-wbl2 wait otherwise) -> relaxed ticket;
  // consumer: acquire fence -> barrier -> plain loads.
  __shared__ unsigned bn_last;
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    const unsigned prev = __hip_atomic_fetch_add(
        &murmura_bn_ticket, 1u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    bn_last = (prev == gridDim.x - 1) ? 1u : 0u;
    if (bn_last) {
      __hip_atomic_store(&murmura_bn_ticket, 0u, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
    }
  }
  __syncthreads();
  if (!bn_last) return;
  if (threadIdx.x == 0) __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  __syncthreads();
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    float s, q;
    bn_reduce_partials(ws, gridDim.x, C, c, s, q);
    if (!BWD) {
      const float m = s / (float)R;
      const float var = fmaxf(q / (float)R - m * m, 0.0f);
      const float inv = rsqrtf(var + eps);
      const float sc = inv * (w ? to_f(w[c]) : 1.0f);
      coef[c] = sc;
      coef[C + c] = (b ? to_f(b[c]) : 0.0f) - m * sc;
      save_mean[c] = m;
      save_invstd[c] = inv;
      if (running_mean != nullptr) {
        from_f(running_mean[c],
               (1.0f - momentum) * to_f(running_mean[c]) + momentum * m);
        const float unbiased = R > 1 ? var * (float)R / (float)(R - 1) : var;
        from_f(running_var[c],
               (1.0f - momentum) * to_f(running_var[c]) + momentum * unbiased);
      }
    } else {
      coef[c] = (w ? to_f(w[c]) : 1.0f) * invstd[c];  // g_scale
      coef[C + c] = s / (float)R;                     // g_mean
      coef[2 * C + c] = q / (float)R;                 // g_proj
      from_f(dweight[c], q);
      from_f(dbias[c], s);
    }
  }
}

// finalize (fwd): reduce G partials -> mean/invstd, persist saved stats,
// EMA-update running stats, precompute scale/shift for the norm pass.
// Threads own channels; reads of ws[g][c] coalesce across threads.
template <typename T>
__global__ void bn_finalize_fwd_kernel(const float* __restrict__ ws, int G,
                                       int64_t R, int C, float eps, float momentum,
                                       const T* __restrict__ w,
                                       const T* __restrict__ b,
                                       float* __restrict__ coef,
                                       float* __restrict__ save_mean,
                                       float* __restrict__ save_invstd,
                                       T* __restrict__ running_mean,
                                       T* __restrict__ running_var) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s, q;
  bn_reduce_partials(ws, G, C, c, s, q);
  float m = s / (float)R;
  float var = fmaxf(q / (float)R - m * m, 0.0f);
  float inv = rsqrtf(var + eps);
  float sc = inv * (w ? to_f(w[c]) : 1.0f);
  coef[c] = sc;
  coef[C + c] = (b ? to_f(b[c]) : 0.0f) - m * sc;
  save_mean[c] = m;
  save_invstd[c] = inv;
  if (running_mean != nullptr) {
    from_f(running_mean[c], (1.0f - momentum) * to_f(running_mean[c]) + momentum * m);
    float unbiased = R > 1 ? var * (float)R / (float)(R - 1) : var;
    from_f(running_var[c], (1.0f - momentum) * to_f(running_var[c]) + momentum * unbiased);
  }
}

// finalize (bwd): reduce G partials -> per-channel backward coefficients
// g_scale = w*invstd, g_mean = sum_dy/R, g_proj = sum_dyx/R; also dweight =
// sum_dyx, dbias = sum_dy.
template <typename T>
__global__ void bn_finalize_bwd_kernel(const float* __restrict__ ws, int G,
                                       int64_t R, int C,
                                       const float* __restrict__ invstd,
                                       const T* __restrict__ w,
                                       float* __restrict__ gcoef,
                                       T* __restrict__ dweight,
                                       T* __restrict__ dbias) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s, q;
  bn_reduce_partials(ws, G, C, c, s, q);
  gcoef[c] = (w ? to_f(w[c]) : 1.0f) * invstd[c];
  gcoef[C + c] = s / (float)R;
  gcoef[2 * C + c] = q / (float)R;
  from_f(dweight[c], q);
  from_f(dbias[c], s);
}

// pass 2 (training): y = [relu](scale*x + shift [+ res]); coefficients come
// precomputed from the finalize kernel (one LDS stage per block, no
// per-block recompute).
template <typename T, bool RELU, bool RES>
__global__ void bn_norm_kernel(const T* __restrict__ x, const T* __restrict__ res,
                               T* __restrict__ y, int64_t R, int C,
                               const float* __restrict__ coef) {
  extern __shared__ float sc[];  // scale[C], shift[C]
  for (int c = threadIdx.x; c < 2 * C; c += blockDim.x) sc[c] = coef[c];
  __syncthreads();
  float* scale = sc;
  float* shift = sc + C;
  constexpr int N = Pack16<T>::N;
  const int64_t nvec = R * C / N;
  const int cvec = C / N;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec; v += stride) {
    Pack16<T> pv = reinterpret_cast<const Pack16<T>*>(x)[v];
    Pack16<T> rv;
    if (RES) rv = reinterpret_cast<const Pack16<T>*>(res)[v];
    const int cbase = (int)(v % cvec) * N;
#pragma unroll
    for (int k = 0; k < N; ++k) {
      float val = fmaf(to_f(pv.e[k]), scale[cbase + k], shift[cbase + k]);
      if (RES) val += to_f(rv.e[k]);
      if (RELU) val = fmaxf(val, 0.0f);
      from_f(pv.e[k], val);
    }
    reinterpret_cast<Pack16<T>*>(y)[v] = pv;
  }
}

// eval-mode forward: same as bn_norm but coefficients from running stats
template <typename T, bool RELU>
__global__ void bn_eval_kernel(const T* __restrict__ x, const T* __restrict__ res,
                               T* __restrict__ y, int64_t R,
                               int C, const T* __restrict__ running_mean,
                               const T* __restrict__ running_var, float eps,
                               const T* __restrict__ w, const T* __restrict__ b,
                               bool has_res) {
  extern __shared__ float coef[];
  float* scale = coef;
  float* shift = coef + C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    float inv = rsqrtf(to_f(running_var[c]) + eps);
    float sc = inv * (w ? to_f(w[c]) : 1.0f);
    scale[c] = sc;
    shift[c] = (b ? to_f(b[c]) : 0.0f) - to_f(running_mean[c]) * sc;
  }
  __syncthreads();
  constexpr int N = Pack16<T>::N;
  const int64_t nvec = R * C / N;
  const int cvec = C / N;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec; v += stride) {
    Pack16<T> pv = reinterpret_cast<const Pack16<T>*>(x)[v];
    Pack16<T> rv;
    if (has_res) rv = reinterpret_cast<const Pack16<T>*>(res)[v];
    const int cbase = (int)(v % cvec) * N;
#pragma unroll
    for (int k = 0; k < N; ++k) {
      float val = fmaf(to_f(pv.e[k]), scale[cbase + k], shift[cbase + k]);
      if (has_res) val += to_f(rv.e[k]);
      if (RELU) val = fmaxf(val, 0.0f);
      from_f(pv.e[k], val);
    }
    reinterpret_cast<Pack16<T>*>(y)[v] = pv;
  }
}

// backward pass 2: dx = g_scale * (g - g_mean - xhat * g_proj) with the
// masked upstream grad g = dy * [yout > 0]; optionally also writes
// dres = g (the residual branch's gradient — the ReLU follows the add, so
// its mask applies to both branches; saves a threshold_backward + an add
// kernel per ResNet block).
template <typename T, bool RELU, bool DRES>
__global__ void bn_bwd_dx_kernel(const T* __restrict__ x, const T* __restrict__ dy,
                                 const T* __restrict__ yout,
                                 T* __restrict__ dx, T* __restrict__ dres,
                                 int64_t R, int C,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 const float* __restrict__ gcoef) {
  extern __shared__ float co[];  // g_scale[C], g_mean[C], g_proj[C], mean[C], invstd[C]
  for (int c = threadIdx.x; c < 3 * C; c += blockDim.x) co[c] = gcoef[c];
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    co[3 * C + c] = mean[c];
    co[4 * C + c] = invstd[c];
  }
  __syncthreads();
  const float* g_scale = co;
  const float* g_mean = co + C;
  const float* g_proj = co + 2 * C;
  const float* mn = co + 3 * C;
  const float* iv = co + 4 * C;
  constexpr int N = Pack16<T>::N;
  const int64_t nvec = R * C / N;
  const int cvec = C / N;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec; v += stride) {
    Pack16<T> xv = reinterpret_cast<const Pack16<T>*>(x)[v];
    Pack16<T> gv = reinterpret_cast<const Pack16<T>*>(dy)[v];
    Pack16<T> yv, rv;
    if (RELU) yv = reinterpret_cast<const Pack16<T>*>(yout)[v];
    const int cbase = (int)(v % cvec) * N;
#pragma unroll
    for (int k = 0; k < N; ++k) {
      const int c = cbase + k;
      float g = to_f(gv.e[k]);
      if (RELU && to_f(yv.e[k]) <= 0.0f) g = 0.0f;
      if (DRES) from_f(rv.e[k], g);
      float xh = (to_f(xv.e[k]) - mn[c]) * iv[c];
      float t = g - g_mean[c] - xh * g_proj[c];
      from_f(gv.e[k], t * g_scale[c]);
    }
    reinterpret_cast<Pack16<T>*>(dx)[v] = gv;
    if (DRES) reinterpret_cast<Pack16<T>*>(dres)[v] = rv;
  }
}

// ============================================================== K15 wrw conv
// 3x3 stride-1 pad-1 NHWC bf16 weight gradient on MFMA (v_mfma_f32_32x32x16_bf16;
// lane mappings verified by ops/hip/mfma_probe.hip).
//
// Why hand-written: MIOpen's chosen igemm wrw kernels use gfx90a-era
// wt32x32x8 tiles (~15-27 us at shapes whose roofline is 2-6 us) and the
// atomic variants add 2-3 SubTensorOp launches each (workspace zero +
// fp32->bf16 cast) — together ~450 us of the flagship's 2.3 ms batch
// (profiles/r02_steady_state.md).
//
// Decomposition: dW[k][ky][kx][c] = sum_{n,y,x} X[n,y+ky-1,x+kx-1,c] *
// dY[n,y,x,k] = 9 GEMMs A^T B with A = X-shifted [rows x C], B = dY
// [rows x K], sharing B. Per block: one [64c x 64k] output tile, ALL 9 taps
// (4 waves x one 32x32 quadrant x 9 accumulators = 144 acc VGPRs,
// __launch_bounds__(256,1)). X is staged per (image, y-chunk) into a
// ZERO-PADDED LDS image [ychunk+2][W+2][64c] so every tap's shifted read is
// in-bounds and border masking costs nothing; dY fragments are loaded once
// per MFMA step and reused by all 9 taps. Image-range splitting (SP blocks
// per tile) writes fp32 partial slabs; the LAST-ARRIVING block per tile
// reduces them in-launch and emits bf16 dW — the agent-scope release/
// acquire + ticket-counter recipe follows cdna_hip_programming.md §5
// (split-K reduction) exactly, including the compiler-hazard asm waits.
// Deterministic (ordered slab reduce — unlike MIOpen's atomic igemm).

typedef __bf16 mfma_bf16x8 __attribute__((ext_vector_type(8)));
typedef float mfma_f32x16 __attribute__((ext_vector_type(16)));

union WrwU8 {
  mfma_bf16x8 v;
  unsigned short u[8];
};

template <int W>
__global__ __launch_bounds__(256, 1) void conv3x3s1_wrw_kernel(
    const bf16raw* __restrict__ x,   // [N][H][W][C] (channels_last)
    const bf16raw* __restrict__ dy,  // [N][H][W][K]
    float* __restrict__ slab,        // [ct][kt][SP][9][64c][64k]
    int N, int H, int C, int K, int SP, int upb, int ug) {
  constexpr int ychunk = (W == 4) ? 4 : 8;
  constexpr int steps = ychunk * W / 16;
  extern __shared__ float lds[];  // shared symbol across TU kernels
  unsigned short* ldsu = reinterpret_cast<unsigned short*>(lds);
  const int xpitch = (W + 2) * 64;
  const int ximg = (ychunk + 2) * xpitch;     // one image-slice of the X image
  const int yimg = ychunk * W * 64;           // one image-slice of the dY image
  unsigned short* ldsx = ldsu;                // [ug][ychunk+2][W+2][64]
  unsigned short* ldsy = ldsu + ug * ximg;    // [ug][ychunk][W][64]

  const int ct = blockIdx.x, kt = blockIdx.y, z = blockIdx.z;
  const int c0 = ct * 64, k0 = kt * 64;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lm = lane & 31, half = lane >> 5;
  const int mc = wave >> 1, nk = wave & 1;  // quadrant: c-half, k-half

  mfma_f32x16 acc[9];
#pragma unroll
  for (int t = 0; t < 9; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) acc[t][r] = 0.0f;

  const int ychunks_per_img = H / ychunk;
  const int units = (N / ug) * ychunks_per_img;  // unit = (image group, y group)
  const int u_lo = z * upb;
  const int u_hi = min(units, u_lo + upb);

  for (int u = u_lo; u < u_hi; ++u) {
    const int n0 = (u / ychunks_per_img) * ug;
    const int yg = (u % ychunks_per_img) * ychunk;
    // ---- cooperative LDS fill, two-phase (loads buffered 8 deep so the
    // ds_writes do not force a vmcnt wait per element; loads are issued
    // UNCONDITIONALLY from a clamped address and zeroed by select, per the
    // guide trap: conditional loads de-pipeline into per-element waits)
    {
      const int xtotal = ug * (ychunk + 2) * (W + 2) * 32;
      unsigned buf[8];
      int tt[8];
      int nb = 0;
      for (int t = threadIdx.x; t < xtotal; t += 256) {
        const int cpair = t & 31;
        const int rest = t >> 5;
        const int xx = rest % (W + 2);
        const int rest2 = rest / (W + 2);
        const int yy = rest2 % (ychunk + 2);
        const int img = rest2 / (ychunk + 2);
        const int gy = yg + yy - 1, gx = xx - 1;
        const bool valid = (unsigned)gy < (unsigned)H && (unsigned)gx < (unsigned)W;
        const int64_t off = valid
            ? ((((int64_t)(n0 + img) * H + gy) * W + gx) * C + c0 + cpair * 2)
            : (int64_t)0;
        unsigned v = *reinterpret_cast<const unsigned*>(x + off);
        buf[nb] = valid ? v : 0u;
        tt[nb] = t;
        if (++nb == 8) {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            reinterpret_cast<unsigned*>(ldsx)[tt[j]] = buf[j];
          nb = 0;
        }
      }
      for (int j = 0; j < nb; ++j)
        reinterpret_cast<unsigned*>(ldsx)[tt[j]] = buf[j];
    }
    {
      const int ytotal = ug * ychunk * W * 32;
      unsigned buf[8];
      int tt[8];
      int nb = 0;
      for (int t = threadIdx.x; t < ytotal; t += 256) {
        const int cpair = t & 31;
        const int rest = t >> 5;
        const int xx = rest % W;
        const int rest2 = rest / W;
        const int yy = rest2 % ychunk;
        const int img = rest2 / ychunk;
        buf[nb] = *reinterpret_cast<const unsigned*>(
            dy + ((((int64_t)(n0 + img) * H + yg + yy) * W + xx) * K + k0 + cpair * 2));
        tt[nb] = t;
        if (++nb == 8) {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            reinterpret_cast<unsigned*>(ldsy)[tt[j]] = buf[j];
          nb = 0;
        }
      }
      for (int j = 0; j < nb; ++j)
        reinterpret_cast<unsigned*>(ldsy)[tt[j]] = buf[j];
    }
    __syncthreads();
    // ---- MFMA over 16-pixel groups, all images of the chunk
    for (int img = 0; img < ug; ++img) {
      const unsigned short* xi = ldsx + img * ximg;
      const unsigned short* yi = ldsy + img * yimg;
      for (int s = 0; s < steps; ++s) {
        WrwU8 b;
        int offs[8];
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          const int p = s * 16 + half * 8 + i;
          b.u[i] = yi[p * 64 + nk * 32 + lm];
          const int yy = p / W, xx = p % W;
          offs[i] = ((yy + 1) * (W + 2) + (xx + 1)) * 64 + mc * 32 + lm;
        }
#pragma unroll
        for (int tap = 0; tap < 9; ++tap) {
          const int doff = ((tap / 3 - 1) * (W + 2) + (tap % 3 - 1)) * 64;
          WrwU8 a;
#pragma unroll
          for (int i = 0; i < 8; ++i) a.u[i] = xi[offs[i] + doff];
          acc[tap] =
              __builtin_amdgcn_mfma_f32_32x32x16_bf16(a.v, b.v, acc[tap], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // ---- write this block's fp32 partial slab (coalesced; a separate
  // full-chip reduce kernel combines the SP slabs deterministically)
  const int tile = ct * gridDim.y + kt;
  float* myslab = slab + ((int64_t)tile * SP + z) * (9 * 64 * 64);
#pragma unroll
  for (int tap = 0; tap < 9; ++tap) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * half + mc * 32;  // c
      const int col = lm + nk * 32;                                 // k
      myslab[(tap * 64 + row) * 64 + col] = acc[tap][r];
    }
  }
}

// v3 (W >= 8): transposed, dx-pre-shifted LDS planes. The v2 kernel's A
// fragments were 8 scalar ds_read_u16 gathers per tap per step (72/step) —
// latency-bound at the kernel's 1-wave/SIMD occupancy. Here X is staged as
// THREE dx-baked zero-bordered planes in [c][y][x] (x-major) layout and dY
// transposed to [k][y][x], so every A/B fragment is ONE 16-B-aligned
// ds_read_b128: 10 LDS reads per step instead of 80.
template <int W>
__global__ __launch_bounds__(256, 1) void conv3x3s1_wrw_t_kernel(
    const bf16raw* __restrict__ x,   // [N][H][W][C] (channels_last)
    const bf16raw* __restrict__ dy,  // [N][H][W][K]
    float* __restrict__ slab,        // [ct][kt][SP][9][64c][64k]
    int N, int H, int C, int K, int SP, int upb) {
  constexpr int ychunk = (W == 32) ? 4 : 8;
  constexpr int steps = ychunk * W / 16;
  // +8-short pad keeps every fragment 16-B aligned while breaking the
  // power-of-two dword cycle between lanes' planes (unpadded: 64-way LDS
  // write conflicts and 32-way b128 read conflicts, measured)
  constexpr int PLANE = (ychunk + 2) * W + 8;
  constexpr int YPLANE = ychunk * W + 8;
  extern __shared__ float lds[];
  unsigned short* ldsu = reinterpret_cast<unsigned short*>(lds);
  unsigned short* ldsx = ldsu;                  // [3][64][PLANE]
  unsigned short* ldsy = ldsu + 3 * 64 * PLANE; // [64][YPLANE]

  const int ct = blockIdx.x, kt = blockIdx.y, z = blockIdx.z;
  const int c0 = ct * 64, k0 = kt * 64;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lm = lane & 31, half = lane >> 5;
  const int mc = wave >> 1, nk = wave & 1;

  mfma_f32x16 acc[9];
#pragma unroll
  for (int t = 0; t < 9; ++t)
#pragma unroll
    for (int r = 0; r < 16; ++r) acc[t][r] = 0.0f;

  const int ychunks_per_img = H / ychunk;
  const int units = N * ychunks_per_img;
  const int u_lo = z * upb;
  const int u_hi = min(units, u_lo + upb);

  for (int u = u_lo; u < u_hi; ++u) {
    const int n = u / ychunks_per_img;
    const int yg = (u % ychunks_per_img) * ychunk;
    // zero ONLY the never-sourced entries: the x=0 column of the d=0 copy
    // and the x=W-1 column of the d=2 copy (y-halo rows are written as
    // zeros by the OOB-clamped loads below); a full-region zero pass was
    // ~73 KB of LDS stores per unit — measured waste
    for (int t = threadIdx.x; t < 2 * 64 * (ychunk + 2); t += 256) {
      const int half2 = t / (64 * (ychunk + 2));       // 0: d=0, 1: d=2
      const int rest = t % (64 * (ychunk + 2));
      const int c = rest / (ychunk + 2), yy = rest % (ychunk + 2);
      const int d = half2 * 2;
      const int xcol = half2 ? (W - 1) : 0;
      ldsx[(d * 64 + c) * PLANE + yy * W + xcol] = 0;
    }
    __syncthreads();
    // X: src-driven transposed fill with the dx shift baked per copy
    {
      const int xtotal = (ychunk + 2) * W * 32;  // 4-B loads (2 channels)
      for (int t = threadIdx.x; t < xtotal; t += 256) {
        const int cpair = t & 31;
        const int rest = t >> 5;
        const int gx = rest % W, yy = rest / W;
        const int gy = yg + yy - 1;
        const bool valid = (unsigned)gy < (unsigned)H;
        const int64_t off = valid
            ? ((((int64_t)n * H + gy) * W + gx) * C + c0 + cpair * 2)
            : (int64_t)0;
        unsigned v = *reinterpret_cast<const unsigned*>(x + off);
        if (!valid) v = 0u;
        const unsigned short lo = (unsigned short)(v & 0xFFFFu);
        const unsigned short hi = (unsigned short)(v >> 16);
#pragma unroll
        for (int d = 0; d < 3; ++d) {
          const int xd = gx + 1 - d;
          if ((unsigned)xd < (unsigned)W) {
            const int base = yy * W + xd;
            ldsx[(d * 64 + cpair * 2) * PLANE + base] = lo;
            ldsx[(d * 64 + cpair * 2 + 1) * PLANE + base] = hi;
          }
        }
      }
    }
    // dY: transposed fill [k][y][x]
    {
      const int ytotal = ychunk * W * 32;
      for (int t = threadIdx.x; t < ytotal; t += 256) {
        const int cpair = t & 31;
        const int rest = t >> 5;
        const int gx = rest % W, yy = rest / W;
        unsigned v = *reinterpret_cast<const unsigned*>(
            dy + ((((int64_t)n * H + yg + yy) * W + gx) * K + k0 + cpair * 2));
        const int base = yy * W + gx;
        ldsy[(cpair * 2) * YPLANE + base] = (unsigned short)(v & 0xFFFFu);
        ldsy[(cpair * 2 + 1) * YPLANE + base] = (unsigned short)(v >> 16);
      }
    }
    __syncthreads();
    for (int s = 0; s < steps; ++s) {
      // 16 px of this step: row(s) and x-origins per lane half
      int yrow, x0;
      if (W == 32) {
        yrow = s / 2;
        x0 = (s % 2) * 16 + half * 8;
      } else if (W == 16) {
        yrow = s;
        x0 = half * 8;
      } else {  // W == 8: each half covers one full row
        yrow = s * 2 + half;
        x0 = 0;
      }
      const int kcol = nk * 32 + lm;
      mfma_bf16x8 b = *reinterpret_cast<const mfma_bf16x8*>(
          &ldsy[kcol * YPLANE + yrow * W + x0]);
      const int crow = mc * 32 + lm;
#pragma unroll
      for (int tap = 0; tap < 9; ++tap) {
        const int dy_ = tap / 3 - 1, d = tap % 3;
        const mfma_bf16x8 a = *reinterpret_cast<const mfma_bf16x8*>(
            &ldsx[(d * 64 + crow) * PLANE + (yrow + dy_ + 1) * W + x0]);
        acc[tap] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc[tap], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  const int tile = ct * gridDim.y + kt;
  float* myslab = slab + ((int64_t)tile * SP + z) * (9 * 64 * 64);
#pragma unroll
  for (int tap = 0; tap < 9; ++tap) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * half + mc * 32;
      const int col = lm + nk * 32;
      myslab[(tap * 64 + row) * 64 + col] = acc[tap][r];
    }
  }
}

// reduce kernel: sum the SP slabs of each tile -> bf16 dW [K][3][3][C].
// grid = (tiles*9, 4): block handles one (tap-tile, 16-wide k quarter);
// slab reads are k-contiguous (coalesced), dW writes staged through LDS so
// they are c-contiguous (channels_last weight layout).
__global__ void conv3x3_wrw_reduce_kernel(const float* __restrict__ slab,
                                          bf16raw* __restrict__ dw, int C, int K,
                                          int SP, int kt_dim, int kw) {
  extern __shared__ float lds[];  // [64c][kw]
  const int tt = blockIdx.x;      // tile*9 + tap
  const int tile = tt / 9, tap = tt % 9;
  const int ct = tile / kt_dim, kt = tile % kt_dim;
  const int kq = blockIdx.y;      // k slice of width kw
  const int c0 = ct * 64, k0 = kt * 64 + kq * kw;
  const float* base = slab + ((int64_t)tile * SP * 9 + tap) * (64 * 64) + kq * kw;
  for (int t = threadIdx.x; t < 64 * kw; t += blockDim.x) {
    const int c = t / kw, k = t % kw;
    const float* p = base + c * 64 + k;
    float s0 = 0.0f, s1 = 0.0f, s2 = 0.0f, s3 = 0.0f;
    int sp = 0;
    for (; sp + 4 <= SP; sp += 4) {
      s0 += p[(int64_t)(sp + 0) * 9 * 64 * 64];
      s1 += p[(int64_t)(sp + 1) * 9 * 64 * 64];
      s2 += p[(int64_t)(sp + 2) * 9 * 64 * 64];
      s3 += p[(int64_t)(sp + 3) * 9 * 64 * 64];
    }
    for (; sp < SP; ++sp) s0 += p[(int64_t)sp * 9 * 64 * 64];
    lds[c * kw + k] = (s0 + s1) + (s2 + s3);
  }
  __syncthreads();
  for (int t = threadIdx.x; t < 64 * kw; t += blockDim.x) {
    const int k = t >> 6, c = t & 63;  // c-contiguous writes
    from_f(dw[((int64_t)(k0 + k) * 9 + tap) * C + c0 + c], lds[c * kw + k]);
  }
}

// =================================================================
// bindings
// =================================================================

namespace {

using torch::Tensor;

inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_flat(const Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == at::kFloat || t.scalar_type() == at::kBFloat16,
              name, " must be fp32 or bf16");
}

#define DISPATCH_FT(TENSOR, ...)                        \
  if ((TENSOR).scalar_type() == at::kFloat) {            \
    using elem_t = float;                                \
    __VA_ARGS__;                                         \
  } else {                                               \
    using elem_t = bf16raw;                              \
    __VA_ARGS__;                                         \
  }

Tensor weighted_sum(Tensor stacked, Tensor w, c10::optional<Tensor> out_opt) {
  check_flat(stacked, "stacked");
  TORCH_CHECK(stacked.dim() == 2, "stacked must be [m, P]");
  int m = (int)stacked.size(0);
  int64_t P = stacked.size(1);
  TORCH_CHECK(m <= 256, "weighted_sum: m must be <= 256");
  Tensor wf = w.to(stacked.device(), at::kFloat).contiguous();
  Tensor out = out_opt.has_value() ? *out_opt : at::empty({P}, stacked.options());
  TORCH_CHECK(out.is_contiguous() && out.numel() == P);
  int blocks = grid_for(P / 4, BLOCK);
  DISPATCH_FT(stacked, {
    weighted_sum_kernel<elem_t><<<blocks, BLOCK, 0, cur_stream()>>>(
        (const elem_t*)stacked.data_ptr(), wf.data_ptr<float>(),
        (elem_t*)out.data_ptr(), m, P);
  });
  return out;
}

template <typename elem_t, int M>
void launch_gram(const Tensor& x, Tensor& gram_flat, int64_t P, int64_t ld) {
  int blocks = grid_for(P / 4, BLOCK, 2048);
  gram_kernel<elem_t, M><<<blocks, BLOCK, 0, cur_stream()>>>(
      (const elem_t*)x.data_ptr(), P, ld, gram_flat.data_ptr<float>());
}

// tile-pair dispatch for the decomposed path: sizes are 8 or the ragged
// remainder r = m % 8; only (8,8), (8,r) and (r,r) combinations occur.
template <typename elem_t>
void launch_gram_cross(const elem_t* xi, const elem_t* xj, int mi, int mj,
                       int64_t P, int64_t ld, float* out, int ldo) {
  int blocks = grid_for(P / 4, BLOCK, 2048);
  hipStream_t st = cur_stream();
#define CROSS_CASE(MI, MJ)                                                       \
  if (mi == MI && mj == MJ) {                                                    \
    gram_cross_kernel<elem_t, MI, MJ>                                            \
        <<<blocks, BLOCK, 0, st>>>(xi, xj, P, ld, out, ldo);                     \
    return;                                                                      \
  }
  CROSS_CASE(8, 8)
  CROSS_CASE(8, 1) CROSS_CASE(8, 2) CROSS_CASE(8, 3) CROSS_CASE(8, 4)
  CROSS_CASE(8, 5) CROSS_CASE(8, 6) CROSS_CASE(8, 7)
  CROSS_CASE(1, 1) CROSS_CASE(2, 2) CROSS_CASE(3, 3) CROSS_CASE(4, 4)
  CROSS_CASE(5, 5) CROSS_CASE(6, 6) CROSS_CASE(7, 7)
#undef CROSS_CASE
  TORCH_CHECK(false, "gram_cross: unsupported tile sizes ", mi, "x", mj);
}

// Full [m, m] Gram matrix of the rows of x (stride(1) must be 1; stride(0)
// may exceed size(1) — chunked column views of a [m, P] buffer are accepted,
// which is what lets exchange overlap Gram accumulation with in-flight
// chunks).
Tensor gram(Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.stride(1) == 1,
              "gram: [m, n] cuda with unit inner stride");
  TORCH_CHECK(x.scalar_type() == at::kFloat || x.scalar_type() == at::kBFloat16);
  int m = (int)x.size(0);
  int64_t P = x.size(1);
  int64_t ld = x.stride(0);
  if (m > 64) {
    Tensor xf = x.to(at::kFloat);
    return at::matmul(xf, xf.t());
  }
  if (m > 16) {
    Tensor g = at::zeros({m, m}, x.options().dtype(at::kFloat));
    float* gp = g.data_ptr<float>();
    int ntiles = (m + 7) / 8;
    DISPATCH_FT(x, {
      const elem_t* base = (const elem_t*)x.data_ptr();
      for (int a = 0; a < ntiles; ++a) {
        int mi = std::min(8, m - a * 8);
        for (int b = a; b < ntiles; ++b) {
          int mj = std::min(8, m - b * 8);
          launch_gram_cross<elem_t>(base + (int64_t)a * 8 * ld,
                                    base + (int64_t)b * 8 * ld, mi, mj, P, ld,
                                    gp + (int64_t)a * 8 * m + b * 8, m);
        }
      }
    });
    return g.triu() + g.triu(1).t();
  }
  int npair = m * (m + 1) / 2;
  Tensor gram_flat = at::zeros({npair}, x.options().dtype(at::kFloat));
  DISPATCH_FT(x, {
    switch (m) {
      case 1: launch_gram<elem_t, 1>(x, gram_flat, P, ld); break;
      case 2: launch_gram<elem_t, 2>(x, gram_flat, P, ld); break;
      case 3: launch_gram<elem_t, 3>(x, gram_flat, P, ld); break;
      case 4: launch_gram<elem_t, 4>(x, gram_flat, P, ld); break;
      case 5: launch_gram<elem_t, 5>(x, gram_flat, P, ld); break;
      case 6: launch_gram<elem_t, 6>(x, gram_flat, P, ld); break;
      case 7: launch_gram<elem_t, 7>(x, gram_flat, P, ld); break;
      case 8: launch_gram<elem_t, 8>(x, gram_flat, P, ld); break;
      case 9: launch_gram<elem_t, 9>(x, gram_flat, P, ld); break;
      case 10: launch_gram<elem_t, 10>(x, gram_flat, P, ld); break;
      case 11: launch_gram<elem_t, 11>(x, gram_flat, P, ld); break;
      case 12: launch_gram<elem_t, 12>(x, gram_flat, P, ld); break;
      case 13: launch_gram<elem_t, 13>(x, gram_flat, P, ld); break;
      case 14: launch_gram<elem_t, 14>(x, gram_flat, P, ld); break;
      case 15: launch_gram<elem_t, 15>(x, gram_flat, P, ld); break;
      case 16: launch_gram<elem_t, 16>(x, gram_flat, P, ld); break;
    }
  });
  Tensor g = at::zeros({m, m}, gram_flat.options());
  auto idx = at::triu_indices(m, m, 0, gram_flat.options().dtype(at::kLong));
  g.index_put_({idx[0], idx[1]}, gram_flat);
  return g.triu() + g.triu(1).t();
}

Tensor pairwise_sq_dists(Tensor stacked) {
  check_flat(stacked, "stacked");
  TORCH_CHECK(stacked.dim() == 2);
  int m = (int)stacked.size(0);
  int64_t P = stacked.size(1);
  if (m > 16) {
    Tensor full = gram(stacked);
    Tensor sq = full.diagonal();
    return (sq.unsqueeze(0) + sq.unsqueeze(1) - 2.0 * full).clamp_min_(0.0);
  }
  int npair = m * (m + 1) / 2;
  Tensor gram_flat = at::zeros({npair}, stacked.options().dtype(at::kFloat));
  DISPATCH_FT(stacked, {
    switch (m) {
      case 1: launch_gram<elem_t, 1>(stacked, gram_flat, P, P); break;
      case 2: launch_gram<elem_t, 2>(stacked, gram_flat, P, P); break;
      case 3: launch_gram<elem_t, 3>(stacked, gram_flat, P, P); break;
      case 4: launch_gram<elem_t, 4>(stacked, gram_flat, P, P); break;
      case 5: launch_gram<elem_t, 5>(stacked, gram_flat, P, P); break;
      case 6: launch_gram<elem_t, 6>(stacked, gram_flat, P, P); break;
      case 7: launch_gram<elem_t, 7>(stacked, gram_flat, P, P); break;
      case 8: launch_gram<elem_t, 8>(stacked, gram_flat, P, P); break;
      case 9: launch_gram<elem_t, 9>(stacked, gram_flat, P, P); break;
      case 10: launch_gram<elem_t, 10>(stacked, gram_flat, P, P); break;
      case 11: launch_gram<elem_t, 11>(stacked, gram_flat, P, P); break;
      case 12: launch_gram<elem_t, 12>(stacked, gram_flat, P, P); break;
      case 13: launch_gram<elem_t, 13>(stacked, gram_flat, P, P); break;
      case 14: launch_gram<elem_t, 14>(stacked, gram_flat, P, P); break;
      case 15: launch_gram<elem_t, 15>(stacked, gram_flat, P, P); break;
      case 16: launch_gram<elem_t, 16>(stacked, gram_flat, P, P); break;
    }
  });
  // unpack upper-triangular flat gram -> full [m, m] sq-dist matrix (tiny)
  Tensor gram = at::zeros({m, m}, gram_flat.options());
  auto idx = at::triu_indices(m, m, 0, gram_flat.options().dtype(at::kLong));
  gram.index_put_({idx[0], idx[1]}, gram_flat);
  gram = gram + gram.t() - at::diag(gram.diagonal());
  Tensor sq = gram.diagonal();
  return (sq.unsqueeze(0) + sq.unsqueeze(1) - 2.0 * gram).clamp_min_(0.0);
}

Tensor row_norms(Tensor stacked) {
  check_flat(stacked, "stacked");
  TORCH_CHECK(stacked.dim() == 2);
  int m = (int)stacked.size(0);
  int64_t P = stacked.size(1);
  Tensor out = at::zeros({m}, stacked.options().dtype(at::kFloat));
  dim3 grid(grid_for(P / 4, BLOCK, 1024), m);
  DISPATCH_FT(stacked, {
    row_sqnorm_kernel<elem_t><<<grid, BLOCK, 0, cur_stream()>>>(
        (const elem_t*)stacked.data_ptr(), P, out.data_ptr<float>());
  });
  return out.sqrt_();
}

Tensor l2_dists_to(Tensor own, Tensor stacked) {
  check_flat(stacked, "stacked");
  check_flat(own, "own");
  TORCH_CHECK(stacked.dim() == 2 && own.numel() == stacked.size(1));
  TORCH_CHECK(own.scalar_type() == stacked.scalar_type());
  int k = (int)stacked.size(0);
  int64_t P = stacked.size(1);
  Tensor out = at::zeros({k}, stacked.options().dtype(at::kFloat));
  dim3 grid(grid_for(P / 4, BLOCK, 1024), k);
  DISPATCH_FT(stacked, {
    sqdist_to_kernel<elem_t><<<grid, BLOCK, 0, cur_stream()>>>(
        (const elem_t*)own.data_ptr(), (const elem_t*)stacked.data_ptr(), P,
        out.data_ptr<float>());
  });
  return out.clamp_min_(0.0).sqrt_();
}

Tensor count_sketch_g8(Tensor stacked, Tensor gt, Tensor tailt, int64_t S) {
  check_flat(stacked, "stacked");
  TORCH_CHECK(stacked.dim() == 2);
  int m = (int)stacked.size(0);
  int64_t P = stacked.size(1);
  TORCH_CHECK(S < (1 << 16), "count_sketch_g8: S must fit 16 bits");
  TORCH_CHECK(gt.scalar_type() == at::kInt && tailt.scalar_type() == at::kInt);
  Tensor g = gt.is_cuda() ? gt.contiguous() : gt.to(stacked.device()).contiguous();
  Tensor tt = tailt.is_cuda() ? tailt.contiguous()
                              : tailt.to(stacked.device()).contiguous();
  int64_t ngroups = g.numel();
  int ntail = (int)tt.numel();
  TORCH_CHECK(ngroups * 8 + ntail == P, "count_sketch_g8: table size mismatch");
  Tensor out = at::zeros({m, S}, stacked.options().dtype(at::kFloat));
  size_t one = (size_t)m * S * sizeof(float);
  TORCH_CHECK(one <= 64 * 1024, "count_sketch_g8: m*S too large for LDS");
  int R = std::max(1, std::min(4, (int)((64 * 1024) / one)));
  int blocks = grid_for(ngroups, BLOCK, 2048);
  bool aligned = (P % 8) == 0;
  DISPATCH_FT(stacked, {
    if (aligned) {
      count_sketch_g8_kernel<elem_t, true><<<blocks, BLOCK, one * R, cur_stream()>>>(
          (const elem_t*)stacked.data_ptr(), g.data_ptr<int>(), m, P, ngroups,
          (int)S, tt.data_ptr<int>(), ntail, out.data_ptr<float>(), R);
    } else {
      count_sketch_g8_kernel<elem_t, false><<<blocks, BLOCK, one * R, cur_stream()>>>(
          (const elem_t*)stacked.data_ptr(), g.data_ptr<int>(), m, P, ngroups,
          (int)S, tt.data_ptr<int>(), ntail, out.data_ptr<float>(), R);
    }
  });
  return out;
}

Tensor count_sketch(Tensor stacked, Tensor packed, int64_t S) {
  check_flat(stacked, "stacked");
  TORCH_CHECK(stacked.dim() == 2);
  int m = (int)stacked.size(0);
  int64_t P = stacked.size(1);
  TORCH_CHECK(S * sizeof(float) <= 160 * 1024 - 1024, "sketch too large for LDS");
  TORCH_CHECK(packed.scalar_type() == at::kInt, "count_sketch: packed int32 table");
  Tensor pt = packed.is_cuda() ? packed.contiguous()
                               : packed.to(stacked.device()).contiguous();
  TORCH_CHECK(pt.numel() == P);
  Tensor out = at::zeros({m, S}, stacked.options().dtype(at::kFloat));
  size_t one = (size_t)m * S * sizeof(float);
  if (one <= 64 * 1024) {
    int R = std::max(1, std::min(4, (int)((64 * 1024) / one)));
    int blocks = grid_for(P, BLOCK, 2048);
    DISPATCH_FT(stacked, {
      count_sketch_multirow_kernel<elem_t><<<blocks, BLOCK, one * R, cur_stream()>>>(
          (const elem_t*)stacked.data_ptr(), pt.data_ptr<int>(), m, P, (int)S,
          out.data_ptr<float>(), R);
    });
    return out;
  }
  dim3 grid(grid_for(P, BLOCK, 1024), m);
  size_t lds = (size_t)S * sizeof(float);
  DISPATCH_FT(stacked, {
    count_sketch_kernel<elem_t><<<grid, BLOCK, lds, cur_stream()>>>(
        (const elem_t*)stacked.data_ptr(), pt.data_ptr<int>(), P, (int)S,
        out.data_ptr<float>());
  });
  return out;
}

void sgd_step(Tensor p, Tensor g, double lr) {
  check_flat(p, "p");
  check_flat(g, "g");
  TORCH_CHECK(p.numel() == g.numel() && p.scalar_type() == g.scalar_type());
  int64_t P = p.numel();
  int blocks = grid_for(P / 4, BLOCK);
  DISPATCH_FT(p, {
    sgd_step_kernel<elem_t><<<blocks, BLOCK, 0, cur_stream()>>>(
        (elem_t*)p.data_ptr(), (const elem_t*)g.data_ptr(), (float)lr, P);
  });
}

void sgd_step_lrt(Tensor p, Tensor g, Tensor lr) {
  check_flat(p, "p");
  check_flat(g, "g");
  TORCH_CHECK(p.numel() == g.numel() && p.scalar_type() == g.scalar_type());
  TORCH_CHECK(lr.is_cuda() && lr.scalar_type() == at::kFloat && lr.numel() == 1);
  int64_t P = p.numel();
  int blocks = grid_for(P / 4, BLOCK);
  DISPATCH_FT(p, {
    sgd_step_lrt_kernel<elem_t><<<blocks, BLOCK, 0, cur_stream()>>>(
        (elem_t*)p.data_ptr(), (const elem_t*)g.data_ptr(), lr.data_ptr<float>(), P);
  });
}

Tensor gaussian_inject(Tensor x, double sigma, int64_t seed, int64_t offset) {
  check_flat(x, "x");
  int64_t P = x.numel();
  Tensor out = at::empty_like(x);
  int blocks = grid_for((P + 3) / 4, BLOCK);
  DISPATCH_FT(x, {
    gaussian_inject_kernel<elem_t><<<blocks, BLOCK, 0, cur_stream()>>>(
        (const elem_t*)x.data_ptr(), (elem_t*)out.data_ptr(), (float)sigma,
        (unsigned long long)seed, (unsigned long long)offset, P);
  });
  return out;
}

Tensor scale_inject(Tensor x, double lam) {
  check_flat(x, "x");
  int64_t P = x.numel();
  Tensor out = at::empty_like(x);
  int blocks = grid_for(P / 4, BLOCK);
  DISPATCH_FT(x, {
    scale_kernel<elem_t><<<blocks, BLOCK, 0, cur_stream()>>>(
        (const elem_t*)x.data_ptr(), (elem_t*)out.data_ptr(), (float)lam, P);
  });
  return out;
}

Tensor ce_loss_acc(Tensor logits, Tensor targets) {
  check_flat(logits, "logits");
  TORCH_CHECK(logits.dim() == 2);
  Tensor tg = targets.to(logits.device(), at::kLong).contiguous();
  int B = (int)logits.size(0), C = (int)logits.size(1);
  Tensor out = at::zeros({2}, logits.options().dtype(at::kFloat));
  int blocks = grid_for((int64_t)B, 4, 2048);
  DISPATCH_FT(logits, {
    ce_loss_acc_kernel<elem_t><<<blocks, BLOCK, 0, cur_stream()>>>(
        (const elem_t*)logits.data_ptr(), tg.data_ptr<int64_t>(), B, C,
        out.data_ptr<float>());
  });
  return out;
}

Tensor evidential_stats(Tensor logits, Tensor targets) {
  check_flat(logits, "logits");
  TORCH_CHECK(logits.dim() == 2);
  Tensor tg = targets.to(logits.device(), at::kLong).contiguous();
  int B = (int)logits.size(0), C = (int)logits.size(1);
  Tensor out = at::zeros({4}, logits.options().dtype(at::kFloat));
  int blocks = grid_for((int64_t)B, 4, 2048);
  DISPATCH_FT(logits, {
    evidential_stats_kernel<elem_t><<<blocks, BLOCK, 0, cur_stream()>>>(
        (const elem_t*)logits.data_ptr(), tg.data_ptr<int64_t>(), B, C,
        out.data_ptr<float>());
  });
  return out;
}

// ---------------------------------------------------------------- K14 bindings
// x: channels_last 4-D tensor (underlying memory [N*H*W, C] with C contiguous)
static int64_t bn_check(const Tensor& x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "bn: x must be 4-D cuda");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "bn: x must be channels_last");
  int C = (int)x.size(1);
  TORCH_CHECK(C % 8 == 0 && C <= 2048, "bn: unsupported channel count ", C);
  return x.numel() / C;
}

// grid for the partials phase: enough blocks to cover the chip for large
// tensors, fewer for small ones so the finalize reduction over G stays tiny
// (~>=32KB of input per block)
static inline int bn_partials_grid(int64_t R, int C, int esize, int rows_per_iter) {
  static const int cap = env_int("MURMURA_BN_GRID_CAP", 160);  // swept 96/160/256 on MI355X
  int64_t by_bytes = (R * C * esize) / 32768;
  int64_t g = std::min<int64_t>((R + rows_per_iter - 1) / rows_per_iter,
                                std::max<int64_t>(1, by_bytes));
  return (int)std::min<int64_t>(std::max<int64_t>(g, 1), cap);
}

struct BnFinalizeArgs {
  float eps = 0.0f, momentum = 0.0f;
  const void* w = nullptr;
  const void* b = nullptr;
  float* coef = nullptr;
  float* save_mean = nullptr;
  float* save_invstd = nullptr;
  void* running_mean = nullptr;
  void* running_var = nullptr;
  void* dweight = nullptr;
  void* dbias = nullptr;
};

template <typename elem_t>
static void bn_partials_launch(const Tensor& x, const Tensor* dy, const Tensor* yout,
                               int64_t R, int C, const float* mean,
                               const float* invstd, Tensor& ws, int G, bool relu,
                               const BnFinalizeArgs& fa) {
  constexpr int N = Pack16<elem_t>::N;
  const int nwaves = BLOCK / WAVE;
  const int ppr = C / N;
  const bool pow2 = (ppr & (ppr - 1)) == 0 && ppr <= WAVE;
  size_t lds = (pow2 ? nwaves * 2 * C : 2 * C) * sizeof(float);
#define BN_FA                                                                  \
  fa.eps, fa.momentum, (const elem_t*)fa.w, (const elem_t*)fa.b, fa.coef,      \
      fa.save_mean, fa.save_invstd, (elem_t*)fa.running_mean,                  \
      (elem_t*)fa.running_var, (elem_t*)fa.dweight, (elem_t*)fa.dbias
  if (dy == nullptr) {
    bn_partials_kernel<elem_t, false, false><<<G, BLOCK, lds, cur_stream()>>>(
        (const elem_t*)x.data_ptr(), nullptr, nullptr, R, C, nullptr, nullptr,
        ws.data_ptr<float>(), BN_FA);
  } else if (relu) {
    bn_partials_kernel<elem_t, true, true><<<G, BLOCK, lds, cur_stream()>>>(
        (const elem_t*)x.data_ptr(), (const elem_t*)dy->data_ptr(),
        (const elem_t*)yout->data_ptr(), R, C, mean, invstd, ws.data_ptr<float>(),
        BN_FA);
  } else {
    bn_partials_kernel<elem_t, true, false><<<G, BLOCK, lds, cur_stream()>>>(
        (const elem_t*)x.data_ptr(), (const elem_t*)dy->data_ptr(), nullptr, R, C,
        mean, invstd, ws.data_ptr<float>(), BN_FA);
  }
#undef BN_FA
}

std::vector<Tensor> bn_fwd_train(Tensor x, c10::optional<Tensor> w, c10::optional<Tensor> b,
                                 c10::optional<Tensor> running_mean,
                                 c10::optional<Tensor> running_var, double momentum,
                                 double eps, bool relu, c10::optional<Tensor> res) {
  int64_t R = bn_check(x);
  int C = (int)x.size(1);
  auto fopt = x.options().dtype(at::kFloat);
  if (running_mean.has_value()) {
    TORCH_CHECK(running_mean->scalar_type() == x.scalar_type(),
                "bn: running stats must match input dtype");
  }
  if (res.has_value()) {
    TORCH_CHECK(res->is_contiguous(at::MemoryFormat::ChannelsLast) &&
                    res->sizes() == x.sizes() && res->scalar_type() == x.scalar_type(),
                "bn: residual must be channels_last, same shape/dtype");
  }
  Tensor saved = at::empty({2, C}, fopt);  // save_mean + save_invstd
  Tensor coef = at::empty({2 * C}, fopt);  // scale + shift
  Tensor y = at::empty_like(x);
  float* sm = saved.data_ptr<float>();
  DISPATCH_FT(x, {
    constexpr int N = Pack16<elem_t>::N;
    const int rows_per_iter = std::max(1, BLOCK / (C / N));
    const int G = bn_partials_grid(R, C, (int)sizeof(elem_t), rows_per_iter);
    Tensor ws = at::empty({G, 2 * C}, fopt);  // partials: no pre-zeroing
    static const bool fused = env_int("MURMURA_BN_FUSED", 0) != 0;
    BnFinalizeArgs fa;
    fa.eps = (float)eps;
    fa.momentum = (float)momentum;
    fa.w = w.has_value() ? w->data_ptr() : nullptr;
    fa.b = b.has_value() ? b->data_ptr() : nullptr;
    fa.coef = coef.data_ptr<float>();
    fa.save_mean = sm;
    fa.save_invstd = sm + C;
    fa.running_mean = running_mean.has_value() ? running_mean->data_ptr() : nullptr;
    fa.running_var = running_var.has_value() ? running_var->data_ptr() : nullptr;
    if (!fused) fa.coef = nullptr;
    bn_partials_launch<elem_t>(x, nullptr, nullptr, R, C, nullptr, nullptr, ws, G,
                               false, fa);
    if (!fused) {
      elem_t* rm = running_mean.has_value() ? (elem_t*)running_mean->data_ptr() : nullptr;
      elem_t* rv = running_var.has_value() ? (elem_t*)running_var->data_ptr() : nullptr;
      bn_finalize_fwd_kernel<elem_t><<<grid_for(C, BLOCK), BLOCK, 0, cur_stream()>>>(
          ws.data_ptr<float>(), G, R, C, (float)eps, (float)momentum,
          w.has_value() ? (const elem_t*)w->data_ptr() : nullptr,
          b.has_value() ? (const elem_t*)b->data_ptr() : nullptr,
          coef.data_ptr<float>(), sm, sm + C, rm, rv);
    }
    int blocks = grid_for(R * C / N, BLOCK);
    size_t lds = 2 * C * sizeof(float);
    const elem_t* rp = res.has_value() ? (const elem_t*)res->data_ptr() : nullptr;
    if (relu && rp) {
      bn_norm_kernel<elem_t, true, true><<<blocks, BLOCK, lds, cur_stream()>>>(
          (const elem_t*)x.data_ptr(), rp, (elem_t*)y.data_ptr(), R, C,
          coef.data_ptr<float>());
    } else if (relu) {
      bn_norm_kernel<elem_t, true, false><<<blocks, BLOCK, lds, cur_stream()>>>(
          (const elem_t*)x.data_ptr(), nullptr, (elem_t*)y.data_ptr(), R, C,
          coef.data_ptr<float>());
    } else if (rp) {
      bn_norm_kernel<elem_t, false, true><<<blocks, BLOCK, lds, cur_stream()>>>(
          (const elem_t*)x.data_ptr(), rp, (elem_t*)y.data_ptr(), R, C,
          coef.data_ptr<float>());
    } else {
      bn_norm_kernel<elem_t, false, false><<<blocks, BLOCK, lds, cur_stream()>>>(
          (const elem_t*)x.data_ptr(), nullptr, (elem_t*)y.data_ptr(), R, C,
          coef.data_ptr<float>());
    }
  });
  return {y, saved[0], saved[1]};
}

Tensor bn_fwd_eval(Tensor x, c10::optional<Tensor> w, c10::optional<Tensor> b,
                   Tensor running_mean, Tensor running_var, double eps, bool relu,
                   c10::optional<Tensor> res) {
  int64_t R = bn_check(x);
  int C = (int)x.size(1);
  Tensor y = at::empty_like(x);
  size_t lds = 2 * C * sizeof(float);
  TORCH_CHECK(running_mean.scalar_type() == x.scalar_type(),
              "bn: running stats must match input dtype");
  DISPATCH_FT(x, {
    constexpr int N = Pack16<elem_t>::N;
    int blocks = grid_for(R * C / N, BLOCK);
    const elem_t* rp = res.has_value() ? (const elem_t*)res->data_ptr() : nullptr;
    if (relu) {
      bn_eval_kernel<elem_t, true><<<blocks, BLOCK, lds, cur_stream()>>>(
          (const elem_t*)x.data_ptr(), rp, (elem_t*)y.data_ptr(), R, C,
          (const elem_t*)running_mean.data_ptr(),
          (const elem_t*)running_var.data_ptr(), (float)eps,
          w.has_value() ? (const elem_t*)w->data_ptr() : nullptr,
          b.has_value() ? (const elem_t*)b->data_ptr() : nullptr, rp != nullptr);
    } else {
      bn_eval_kernel<elem_t, false><<<blocks, BLOCK, lds, cur_stream()>>>(
          (const elem_t*)x.data_ptr(), rp, (elem_t*)y.data_ptr(), R, C,
          (const elem_t*)running_mean.data_ptr(),
          (const elem_t*)running_var.data_ptr(), (float)eps,
          w.has_value() ? (const elem_t*)w->data_ptr() : nullptr,
          b.has_value() ? (const elem_t*)b->data_ptr() : nullptr, rp != nullptr);
    }
  });
  return y;
}

std::vector<Tensor> bn_bwd(Tensor x, Tensor dy, c10::optional<Tensor> w, Tensor mean,
                           Tensor invstd, c10::optional<Tensor> yout, bool relu,
                           bool need_dres) {
  TORCH_CHECK(!relu || yout.has_value(), "bn_bwd: relu needs the saved output");
  TORCH_CHECK(!need_dres || relu, "bn_bwd: residual fusion implies fused relu");
  int64_t R = bn_check(x);
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast));
  int C = (int)x.size(1);
  auto fopt = x.options().dtype(at::kFloat);
  Tensor gcoef = at::empty({3 * C}, fopt);
  Tensor dx = at::empty_like(dy);
  Tensor dres = need_dres ? at::empty_like(dy) : Tensor();
  Tensor dweight = at::empty({C}, x.options());
  Tensor dbias = at::empty({C}, x.options());
  hipStream_t st = cur_stream();
  DISPATCH_FT(x, {
    constexpr int N = Pack16<elem_t>::N;
    const int rows_per_iter = std::max(1, BLOCK / (C / N));
    const int G = bn_partials_grid(R, C, (int)sizeof(elem_t), rows_per_iter);
    Tensor ws = at::empty({G, 2 * C}, fopt);  // partials: no pre-zeroing
    const Tensor* yp = yout.has_value() ? &*yout : nullptr;
    static const bool fused2 = env_int("MURMURA_BN_FUSED", 0) != 0;
    BnFinalizeArgs fa;
    fa.w = w.has_value() ? w->data_ptr() : nullptr;
    fa.coef = fused2 ? gcoef.data_ptr<float>() : nullptr;
    fa.dweight = dweight.data_ptr();
    fa.dbias = dbias.data_ptr();
    bn_partials_launch<elem_t>(x, &dy, yp, R, C, mean.data_ptr<float>(),
                               invstd.data_ptr<float>(), ws, G, relu, fa);
    if (!fused2) {
      bn_finalize_bwd_kernel<elem_t><<<grid_for(C, BLOCK), BLOCK, 0, st>>>(
          ws.data_ptr<float>(), G, R, C, invstd.data_ptr<float>(),
          w.has_value() ? (const elem_t*)w->data_ptr() : nullptr,
          gcoef.data_ptr<float>(), (elem_t*)dweight.data_ptr(),
          (elem_t*)dbias.data_ptr());
    }
    int blocks = grid_for(R * C / N, BLOCK);
    size_t lds = 5 * C * sizeof(float);
    const elem_t* yp2 = yp ? (const elem_t*)yp->data_ptr() : nullptr;
    if (relu && need_dres) {
      bn_bwd_dx_kernel<elem_t, true, true><<<blocks, BLOCK, lds, st>>>(
          (const elem_t*)x.data_ptr(), (const elem_t*)dy.data_ptr(), yp2,
          (elem_t*)dx.data_ptr(), (elem_t*)dres.data_ptr(), R, C,
          mean.data_ptr<float>(), invstd.data_ptr<float>(), gcoef.data_ptr<float>());
    } else if (relu) {
      bn_bwd_dx_kernel<elem_t, true, false><<<blocks, BLOCK, lds, st>>>(
          (const elem_t*)x.data_ptr(), (const elem_t*)dy.data_ptr(), yp2,
          (elem_t*)dx.data_ptr(), nullptr, R, C,
          mean.data_ptr<float>(), invstd.data_ptr<float>(), gcoef.data_ptr<float>());
    } else {
      bn_bwd_dx_kernel<elem_t, false, false><<<blocks, BLOCK, lds, st>>>(
          (const elem_t*)x.data_ptr(), (const elem_t*)dy.data_ptr(), yp2,
          (elem_t*)dx.data_ptr(), nullptr, R, C,
          mean.data_ptr<float>(), invstd.data_ptr<float>(), gcoef.data_ptr<float>());
    }
  });
  if (need_dres) return {dx, dweight, dbias, dres};
  return {dx, dweight, dbias};
}

Tensor conv3x3s1_wrw(Tensor x, Tensor dy) {
  TORCH_CHECK(x.is_cuda() && dy.is_cuda() && x.dim() == 4 && dy.dim() == 4,
              "conv3x3s1_wrw: 4-D cuda tensors");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && dy.scalar_type() == at::kBFloat16,
              "conv3x3s1_wrw: bf16 only");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                  dy.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv3x3s1_wrw: channels_last");
  const int N = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3);
  const int K = (int)dy.size(1);
  TORCH_CHECK(dy.size(0) == N && dy.size(2) == H && dy.size(3) == W,
              "conv3x3s1_wrw: stride-1 same-shape only");
  TORCH_CHECK(C % 64 == 0 && K % 64 == 0 && C <= 512 && K <= 512,
              "conv3x3s1_wrw: C,K multiples of 64 (<=512)");
  TORCH_CHECK(W == 4 || W == 8 || W == 16 || W == 32,
              "conv3x3s1_wrw: W in {4,8,16,32}");
  // W >= 8 uses the transposed v3 kernel (ychunk 4 for W=32, 8 otherwise);
  // W == 4 keeps the v2 padded-row kernel with image grouping
  const bool v3 = W >= 8;
  const int ychunk = v3 ? (W == 32 ? 4 : 8) : std::min(H, 8);
  TORCH_CHECK(H % ychunk == 0, "H not divisible by ychunk");
  int ug = 1;
  if (!v3 && W <= 8) {
    const size_t ximg = (size_t)(ychunk + 2) * (W + 2) * 64 * 2;
    const size_t yimg = (size_t)ychunk * W * 64 * 2;
    while (ug * 2 <= N && (size_t)(ug * 2) * (ximg + yimg) <= 100 * 1024 &&
           ug * 2 * (ychunk * W / 16) <= 32) {
      ug *= 2;
    }
  }
  TORCH_CHECK(N % ug == 0, "batch not divisible by image group");
  const int ct = C / 64, kt = K / 64;
  const int units = (N / ug) * (H / ychunk);
  int SP = std::max(1, std::min(units, 128 / (ct * kt)));
  const int upb = (units + SP - 1) / SP;
  SP = (units + upb - 1) / upb;
  auto fopt = x.options().dtype(at::kFloat);
  Tensor slab = at::empty({(int64_t)ct * kt * SP * 9 * 64 * 64}, fopt);
  Tensor dw = at::empty({K, C, 3, 3},
                        x.options().memory_format(at::MemoryFormat::ChannelsLast));
  const size_t lds = v3
      ? ((size_t)3 * 64 * ((ychunk + 2) * W + 8) +
         (size_t)64 * (ychunk * W + 8)) * sizeof(unsigned short)
      : ((size_t)ug * (ychunk + 2) * (W + 2) * 64 +
         (size_t)ug * ychunk * W * 64) * sizeof(unsigned short);
  static bool attr_set = false;
  if (!attr_set) {
    (void)hipFuncSetAttribute((const void*)conv3x3s1_wrw_kernel<4>,
                              hipFuncAttributeMaxDynamicSharedMemorySize, 144 * 1024);
    (void)hipFuncSetAttribute((const void*)conv3x3s1_wrw_kernel<8>,
                              hipFuncAttributeMaxDynamicSharedMemorySize, 144 * 1024);
    (void)hipFuncSetAttribute((const void*)conv3x3s1_wrw_kernel<16>,
                              hipFuncAttributeMaxDynamicSharedMemorySize, 144 * 1024);
    (void)hipFuncSetAttribute((const void*)conv3x3s1_wrw_kernel<32>,
                              hipFuncAttributeMaxDynamicSharedMemorySize, 144 * 1024);
    (void)hipFuncSetAttribute((const void*)conv3x3s1_wrw_t_kernel<8>,
                              hipFuncAttributeMaxDynamicSharedMemorySize, 144 * 1024);
    (void)hipFuncSetAttribute((const void*)conv3x3s1_wrw_t_kernel<16>,
                              hipFuncAttributeMaxDynamicSharedMemorySize, 144 * 1024);
    (void)hipFuncSetAttribute((const void*)conv3x3s1_wrw_t_kernel<32>,
                              hipFuncAttributeMaxDynamicSharedMemorySize, 144 * 1024);
    attr_set = true;
  }
  dim3 grid(ct, kt, SP);
  if (v3) {
#define WRW_T_LAUNCH(WW)                                                        \
    conv3x3s1_wrw_t_kernel<WW><<<grid, 256, lds, cur_stream()>>>(               \
        (const bf16raw*)x.data_ptr(), (const bf16raw*)dy.data_ptr(),            \
        slab.data_ptr<float>(), N, H, C, K, SP, upb)
    switch (W) {
      case 8: WRW_T_LAUNCH(8); break;
      case 16: WRW_T_LAUNCH(16); break;
      case 32: WRW_T_LAUNCH(32); break;
    }
#undef WRW_T_LAUNCH
  } else {
#define WRW_LAUNCH(WW)                                                          \
  conv3x3s1_wrw_kernel<WW><<<grid, 256, lds, cur_stream()>>>(                   \
      (const bf16raw*)x.data_ptr(), (const bf16raw*)dy.data_ptr(),              \
      slab.data_ptr<float>(), N, H, C, K, SP, upb, ug)
    WRW_LAUNCH(4);
#undef WRW_LAUNCH
  }
  int kw = 16;  // k-slice width: more blocks when SP (slab count) is large
  while (kw > 4 && ct * kt * 9 * (64 / kw) < 128) kw /= 2;
  dim3 rgrid(ct * kt * 9, 64 / kw);
  conv3x3_wrw_reduce_kernel<<<rgrid, 256, 64 * kw * sizeof(float), cur_stream()>>>(
      slab.data_ptr<float>(), (bf16raw*)dw.data_ptr(), C, K, SP, kt, kw);
  return dw;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("bn_fwd_train", &bn_fwd_train,
        "K14: fused NHWC BatchNorm(+ReLU)(+residual) training forward -> "
        "(y, mean, invstd)",
        py::arg("x"), py::arg("w"), py::arg("b"), py::arg("running_mean"),
        py::arg("running_var"), py::arg("momentum"), py::arg("eps"),
        py::arg("relu") = false, py::arg("res") = py::none());
  m.def("bn_fwd_eval", &bn_fwd_eval,
        "K14: NHWC BatchNorm(+ReLU)(+residual) eval forward",
        py::arg("x"), py::arg("w"), py::arg("b"), py::arg("running_mean"),
        py::arg("running_var"), py::arg("eps"), py::arg("relu") = false,
        py::arg("res") = py::none());
  m.def("bn_bwd", &bn_bwd,
        "K14: NHWC BatchNorm(+ReLU)(+residual) backward -> "
        "(dx, dweight, dbias[, dres])",
        py::arg("x"), py::arg("dy"), py::arg("w"), py::arg("mean"), py::arg("invstd"),
        py::arg("yout") = py::none(), py::arg("relu") = false,
        py::arg("need_dres") = false);
  m.def("weighted_sum", &weighted_sum, "K1: out = sum_i w_i * x_i",
        py::arg("stacked"), py::arg("w"), py::arg("out") = py::none());
  m.def("pairwise_sq_dists", &pairwise_sq_dists, "K2: [m,m] squared L2 matrix");
  m.def("gram", &gram, "K2: [m,m] Gram matrix (accepts chunked column views)");
  m.def("row_norms", &row_norms, "K12: per-row L2 norms");
  m.def("l2_dists_to", &l2_dists_to, "K2 variant: dists of rows to own");
  m.def("count_sketch", &count_sketch, "K4: count-sketch projection");
  m.def("count_sketch_g8", &count_sketch_g8,
        "K4: count-sketch with group-of-8 bins (1 atomic per group per row)");
  m.def("conv3x3s1_wrw", &conv3x3s1_wrw,
        "K15: 3x3 s1 p1 NHWC bf16 conv weight gradient (MFMA 32x32x16, "
        "in-launch split-image reduction)");
  m.def("sgd_step", &sgd_step, "K6: fused p -= lr*g");
  m.def("sgd_step_lrt", &sgd_step_lrt, "K6: fused p -= lr*g, lr from device");
  m.def("gaussian_inject", &gaussian_inject, "K10: x + N(0, sigma^2) (Philox)");
  m.def("scale_inject", &scale_inject, "K11: lam * x");
  m.def("ce_loss_acc", &ce_loss_acc, "K7: [loss_sum, correct]");
  m.def("evidential_stats", &evidential_stats,
        "K8: [vacuity_sum, entropy_sum, strength_sum, correct]");
}
