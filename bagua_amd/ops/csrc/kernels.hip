// CDNA4 (gfx950) kernel pack for bagua_amd.
//
// From-scratch MI355X implementations of the device math the reference
// framework shipped as CUDA (reference inventory:
// rust/bagua-core/bagua-core-internal/kernels/bagua_kernels.cu:196-501):
// elementwise bucket ops, chunked in-place reduce, MinMaxUInt8
// compress/decompress.
//
// Design (per the CDNA4 programming guide):
//  * 256-thread workgroups (4 waves of 64);
//  * every kernel is memory-bound -> 16-byte vectorized loads/stores per
//    lane (guide G13: scalar bf16/f16 loads are ~2-2.5x slower);
//  * grid capped at ~2048 workgroups with grid-stride loops (guide G11);
//  * min/max reductions: wave shuffle reduce over 64 lanes, then LDS
//    across the 4 waves, then one ordered-uint atomic per block
//    (guide: Reduction quick reference);
//  * float32 accumulation for f16/bf16 inputs.
//
// Compression wire format (matches bagua_amd/ops/quant.py, the numerics
// oracle): per chunk a 32-byte header whose first 8 bytes are (min,max)
// as float32, then ceil32(chunk) uint8 payload.
//   scale = 255/(max-min+1e-7); upper = rint(max*scale);
//   lower = upper-255; q = min(rint(x*scale), upper) - lower.

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define BLOCK 256
#define MAX_GRID 2048

// ---------------------------------------------------------------------------
// dtype helpers
// ---------------------------------------------------------------------------

template <typename T> __device__ __forceinline__ float to_f(T v);
template <> __device__ __forceinline__ float to_f<float>(float v) { return v; }
template <> __device__ __forceinline__ float to_f<__half>(__half v) {
  return __half2float(v);
}
template <> __device__ __forceinline__ float to_f<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}

template <typename T> __device__ __forceinline__ T from_f(float v);
template <> __device__ __forceinline__ float from_f<float>(float v) { return v; }
template <> __device__ __forceinline__ __half from_f<__half>(float v) {
  return __float2half(v);
}
template <> __device__ __forceinline__ __hip_bfloat16 from_f<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

// 16-byte per-lane vector of T. Access widths are now tuned PER KERNEL:
// elementwise uses 2x Vec16 (32 B), the compressor kernels use a fixed
// 16 u8 payload granule (16 B) with 16*sizeof(T)/16 Vec16 tensor-side
// vectors — the round-1 one-width-fits-all tradeoff is gone.
template <typename T> struct alignas(16) Vec16 {
  static constexpr int N = 16 / sizeof(T);
  T v[N];
};

// ---------------------------------------------------------------------------
// elementwise ops: x = f(x, y)
// ---------------------------------------------------------------------------

enum EwOp { EW_AVERAGE = 0, EW_ADD = 1, EW_SUB = 2, EW_ADDMUL = 3, EW_DIV = 4 };

template <int OP>
__device__ __forceinline__ float ew_apply(float x, float y, float f) {
  if (OP == EW_AVERAGE) return (x + y) * 0.5f;
  if (OP == EW_ADD) return x + y;
  if (OP == EW_SUB) return x - y;
  if (OP == EW_ADDMUL) return x + y * f;
  return x * f;  // EW_DIV passes f = 1/divisor
}

template <typename T, int OP>
__global__ void ew_kernel(T* __restrict__ x, const T* __restrict__ y,
                          float f, size_t n) {
  // two 16-byte vectors per lane per iteration (32 B accesses): +9% on
  // the f32 read-modify-write pattern (round-1 measurement); unlike the
  // compressor kernels this path has no competing access width.
  constexpr int V = Vec16<T>::N;
  using VT = Vec16<T>;
  const size_t tid = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const size_t nv2 = n / (2 * V);

  VT* xv = reinterpret_cast<VT*>(x);
  const VT* yv = reinterpret_cast<const VT*>(y);
  for (size_t i = tid; i < nv2; i += stride) {
    VT a0 = xv[2 * i], a1 = xv[2 * i + 1];
    VT b0, b1;
    if (OP != EW_DIV) {
      b0 = yv[2 * i];
      b1 = yv[2 * i + 1];
    }
#pragma unroll
    for (int k = 0; k < V; ++k) {
      float bx0 = (OP != EW_DIV) ? to_f(b0.v[k]) : 0.f;
      float bx1 = (OP != EW_DIV) ? to_f(b1.v[k]) : 0.f;
      a0.v[k] = from_f<T>(ew_apply<OP>(to_f(a0.v[k]), bx0, f));
      a1.v[k] = from_f<T>(ew_apply<OP>(to_f(a1.v[k]), bx1, f));
    }
    xv[2 * i] = a0;
    xv[2 * i + 1] = a1;
  }
  for (size_t i = nv2 * 2 * V + tid; i < n; i += stride) {
    float bx = (OP != EW_DIV) ? to_f(y[i]) : 0.f;
    x[i] = from_f<T>(ew_apply<OP>(to_f(x[i]), bx, f));
  }
}

// x += reduced / nranks - x_copy  (async model average,
// reference: bagua_kernels.cu:257-267)
template <typename T>
__global__ void async_avg_kernel(T* __restrict__ x,
                                 const T* __restrict__ reduced,
                                 const T* __restrict__ x_copy,
                                 float inv_n, size_t n) {
  constexpr int V = Vec16<T>::N;
  using VT = Vec16<T>;
  const size_t tid = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const size_t nv = n / V;
  VT* xv = reinterpret_cast<VT*>(x);
  const VT* rv = reinterpret_cast<const VT*>(reduced);
  const VT* cv = reinterpret_cast<const VT*>(x_copy);
  for (size_t i = tid; i < nv; i += stride) {
    VT a = xv[i], r = rv[i], c = cv[i];
#pragma unroll
    for (int k = 0; k < V; ++k)
      a.v[k] = from_f<T>(to_f(a.v[k]) + to_f(r.v[k]) * inv_n - to_f(c.v[k]));
    xv[i] = a;
  }
  for (size_t i = nv * V + tid; i < n; i += stride)
    x[i] = from_f<T>(to_f(x[i]) + to_f(reduced[i]) * inv_n - to_f(x_copy[i]));
}

// ---------------------------------------------------------------------------
// chunked in-place reduce: x[target*chunk + i] = reduce_c x[c*chunk + i]
// (reference: bagua_kernels.cu:374-401) — vectorized over i, small loop
// over chunks accumulating in f32.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void reduce_chunk_kernel(T* __restrict__ x, int num_chunks,
                                    int target_chunk, float post_scale,
                                    size_t chunk) {
  constexpr int V = Vec16<T>::N;
  using VT = Vec16<T>;
  const size_t tid = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const size_t nv = chunk / V;
  VT* tv = reinterpret_cast<VT*>(x + (size_t)target_chunk * chunk);
  for (size_t i = tid; i < nv; i += stride) {
    float acc[Vec16<T>::N];
#pragma unroll
    for (int k = 0; k < V; ++k) acc[k] = 0.f;
    for (int c = 0; c < num_chunks; ++c) {
      const VT* cvp = reinterpret_cast<const VT*>(x + (size_t)c * chunk);
      VT b = cvp[i];
#pragma unroll
      for (int k = 0; k < V; ++k) acc[k] += to_f(b.v[k]);
    }
    VT out;
#pragma unroll
    for (int k = 0; k < V; ++k) out.v[k] = from_f<T>(acc[k] * post_scale);
    tv[i] = out;
  }
  for (size_t i = nv * V + tid; i < chunk; i += stride) {
    float acc = 0.f;
    for (int c = 0; c < num_chunks; ++c) acc += to_f(x[(size_t)c * chunk + i]);
    x[(size_t)target_chunk * chunk + i] = from_f<T>(acc * post_scale);
  }
}

// ---------------------------------------------------------------------------
// MinMaxUInt8 compression
// ---------------------------------------------------------------------------

// monotonic float<->uint mapping so integer atomicMin/Max give float min/max
__device__ __forceinline__ uint32_t f32_to_ordered(float f) {
  uint32_t u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}
__device__ __forceinline__ float ordered_to_f32(uint32_t u) {
  u = (u & 0x80000000u) ? (u & 0x7FFFFFFFu) : ~u;
  return __uint_as_float(u);
}

#define MINMAX_MAX_BLOCKS 512

// stage 2: one block per chunk reduces the per-block partials (no
// atomics — 512 blocks contending on one address serialized ~100us)
__global__ void minmax_finalize_kernel(const uint32_t* __restrict__
                                       partials, int nblocks,
                                       int chunk_begin,
                                       uint32_t* __restrict__ scratch) {
  const int c = chunk_begin + blockIdx.x;
  const uint32_t* p = partials + (size_t)blockIdx.x * 2 * MINMAX_MAX_BLOCKS;
  uint32_t mn = 0xFFFFFFFFu, mx = 0u;
  for (int i = threadIdx.x; i < nblocks; i += blockDim.x) {
    mn = min(mn, p[2 * i]);
    mx = max(mx, p[2 * i + 1]);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    mn = min(mn, (uint32_t)__shfl_down((int)mn, off, 64));
    mx = max(mx, (uint32_t)__shfl_down((int)mx, off, 64));
  }
  __shared__ uint32_t smn[BLOCK / 64], smx[BLOCK / 64];
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  if (lane == 0) {
    smn[wave] = mn;
    smx[wave] = mx;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
#pragma unroll
    for (int w = 1; w < BLOCK / 64; ++w) {
      mn = min(mn, smn[w]);
      mx = max(mx, smx[w]);
    }
    scratch[2 * c] = mn;
    scratch[2 * c + 1] = mx;
  }
}

// grid.y = chunk index (offset by chunk_begin); block-level min/max with
// wave shuffle + LDS across 4 waves, one atomic pair per block.
template <typename T>
__global__ void minmax_kernel(const T* __restrict__ x, size_t chunk,
                              int chunk_begin, uint32_t* partials) {
  constexpr int V = Vec16<T>::N;
  using VT = Vec16<T>;
  const int c = chunk_begin + blockIdx.y;
  const T* __restrict__ src = x + (size_t)c * chunk;

  // 4 independent accumulator pairs: a single (lmin,lmax) pair makes a
  // serial FP dependency chain per thread and the kernel goes
  // latency-bound (~1.2 TB/s measured); 4 chains restore ILP.
  float lmn[4] = {INFINITY, INFINITY, INFINITY, INFINITY};
  float lmx[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
  const size_t tid = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const size_t nv = chunk / V;
  const VT* sv = reinterpret_cast<const VT*>(src);
  for (size_t i = tid; i < nv; i += stride) {
    VT a = sv[i];
#pragma unroll
    for (int k = 0; k < V; ++k) {
      float f = to_f(a.v[k]);
      lmn[k & 3] = fminf(lmn[k & 3], f);
      lmx[k & 3] = fmaxf(lmx[k & 3], f);
    }
  }
  for (size_t i = nv * V + tid; i < chunk; i += stride) {
    float f = to_f(src[i]);
    lmn[0] = fminf(lmn[0], f);
    lmx[0] = fmaxf(lmx[0], f);
  }
  float lmin = fminf(fminf(lmn[0], lmn[1]), fminf(lmn[2], lmn[3]));
  float lmax = fmaxf(fmaxf(lmx[0], lmx[1]), fmaxf(lmx[2], lmx[3]));

  // wave64 shuffle reduce
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    lmin = fminf(lmin, __shfl_down(lmin, off, 64));
    lmax = fmaxf(lmax, __shfl_down(lmax, off, 64));
  }
  __shared__ float smin[BLOCK / 64], smax[BLOCK / 64];
  const int wave = threadIdx.x / 64;
  const int lane = threadIdx.x % 64;
  if (lane == 0) {
    smin[wave] = lmin;
    smax[wave] = lmax;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
#pragma unroll
    for (int w = 1; w < BLOCK / 64; ++w) {
      lmin = fminf(lmin, smin[w]);
      lmax = fmaxf(lmax, smax[w]);
    }
    // per-block partial; chunk-row-local block index
    uint32_t* p = partials + (size_t)blockIdx.y * 2 * MINMAX_MAX_BLOCKS;
    p[2 * blockIdx.x] = f32_to_ordered(lmin);
    p[2 * blockIdx.x + 1] = f32_to_ordered(lmax);
  }
}

__device__ __forceinline__ void quant_params(float mn, float mx,
                                             float& scale, float& lower,
                                             float& upper) {
  scale = 255.0f / (mx - mn + 1e-7f);
  upper = rintf(mx * scale);
  lower = upper - 255.0f;
}

// grid.y = chunk; writes 32B header (min,max float32) + payload
template <typename T>
__global__ void quantize_kernel(const T* __restrict__ x, size_t chunk,
                                int chunk_begin, size_t chunk_stride,
                                const uint32_t* __restrict__ scratch,
                                uint8_t* __restrict__ out) {
  const int c = chunk_begin + blockIdx.y;
  const float mn = ordered_to_f32(scratch[2 * c]);
  const float mx = ordered_to_f32(scratch[2 * c + 1]);
  float scale, lower, upper;
  quant_params(mn, mx, scale, lower, upper);

  uint8_t* __restrict__ dst = out + (size_t)c * chunk_stride;
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    float* hdr = reinterpret_cast<float*>(dst);
    hdr[0] = mn;
    hdr[1] = mx;
    // zero the header tail (bytes 8..31): the wire buffer is sent to
    // peers, so uninitialized bytes would leak GPU memory contents and
    // make the wire non-deterministic vs the zero-padded CPU oracle.
#pragma unroll
    for (int k = 2; k < 8; ++k) hdr[k] = 0.0f;
  }
  const T* __restrict__ src = x + (size_t)c * chunk;
  uint8_t* __restrict__ payload = dst + 32;

  // per-T width (measured, gpurun r2c9): 16-byte tensor-side loads win;
  // widening the u8 store granule beyond that costs ~1%
  constexpr int V = Vec16<T>::N;  // 4 (f32) / 8 (f16, bf16)
  using VT = Vec16<T>;
  const size_t tid = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const size_t nv = chunk / V;
  const VT* sv = reinterpret_cast<const VT*>(src);
  for (size_t i = tid; i < nv; i += stride) {
    VT a = sv[i];
    uint8_t q[Vec16<T>::N];
#pragma unroll
    for (int k = 0; k < V; ++k) {
      float level = fminf(rintf(to_f(a.v[k]) * scale), upper);
      q[k] = (uint8_t)(level - lower);
    }
    if (V == 4)
      *reinterpret_cast<uint32_t*>(payload + i * 4) =
          *reinterpret_cast<uint32_t*>(q);
    else
      *reinterpret_cast<uint64_t*>(payload + i * 8) =
          *reinterpret_cast<uint64_t*>(q);
  }
  for (size_t i = nv * V + tid; i < chunk; i += stride) {
    float level = fminf(rintf(to_f(src[i]) * scale), upper);
    payload[i] = (uint8_t)(level - lower);
  }
}

template <typename T>
__global__ void dequantize_kernel(const uint8_t* __restrict__ in,
                                  size_t chunk, int chunk_begin,
                                  size_t chunk_stride, T* __restrict__ x) {
  const int c = chunk_begin + blockIdx.y;
  const uint8_t* __restrict__ src = in + (size_t)c * chunk_stride;
  const float* hdr = reinterpret_cast<const float*>(src);
  const float mn = hdr[0], mx = hdr[1];
  float scale, lower, upper;
  quant_params(mn, mx, scale, lower, upper);
  const float inv_scale = 1.0f / scale;

  const uint8_t* __restrict__ payload = src + 32;
  T* __restrict__ dst = x + (size_t)c * chunk;

  // per-dtype width (measured, gpurun r2c9): f32 wants V=4 (4-byte
  // payload load + one 16-byte store — V=16's four stores per lane cost
  // -43%); f16/bf16 want V=16 (one 16-byte payload load, two 16-byte
  // stores — +21% over V=8)
  constexpr int V = sizeof(T) == 4 ? 4 : 16;
  constexpr int OV = Vec16<T>::N;
  constexpr int NOUT = V / OV;
  using VT = Vec16<T>;
  const size_t tid = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const size_t nv = chunk / V;
  VT* dv = reinterpret_cast<VT*>(dst);
  for (size_t i = tid; i < nv; i += stride) {
    alignas(16) uint8_t q[V];
    if (V == 4)
      *reinterpret_cast<uint32_t*>(q) =
          *reinterpret_cast<const uint32_t*>(payload + i * 4);
    else
      *reinterpret_cast<uint4*>(q) =
          *reinterpret_cast<const uint4*>(payload + i * 16);
#pragma unroll
    for (int o = 0; o < NOUT; ++o) {
      VT out;
#pragma unroll
      for (int k = 0; k < OV; ++k)
        out.v[k] = from_f<T>(((float)q[o * OV + k] + lower) * inv_scale);
      dv[i * NOUT + o] = out;
    }
  }
  for (size_t i = nv * V + tid; i < chunk; i += stride)
    dst[i] = from_f<T>(((float)payload[i] + lower) * inv_scale);
}

// ---------------------------------------------------------------------------
// fused dequantize + chunk reduce (ByteGrad hot path).
//
// After the alltoall, the bucket's non-target chunks exist only as
// inputs to the reduction — materializing them through a full
// decompress pass and re-reading them in reduce_chunk wastes ~2 bucket
// passes of HBM traffic. This kernel reads the wire payloads directly
// (n_chunks x u8), dequantizes in registers and writes ONLY the target
// chunk. Values round through T between dequantize and accumulate so
// the result is BITWISE identical to dequantize_kernel +
// reduce_chunk_kernel (the CPU oracle and the python executor take that
// unfused chain).
// ---------------------------------------------------------------------------

#define DQR_MAX_CHUNKS 64

template <typename T>
__global__ void dequant_reduce_kernel(const uint8_t* __restrict__ in,
                                      size_t chunk, size_t chunk_stride,
                                      int num_chunks, float post_scale,
                                      T* __restrict__ dst /* target base */) {
  // no FMA contraction: for T=float the dequantized value must round
  // exactly as the unfused dequantize kernel's store did, or bitwise
  // parity with decompress+reduce breaks (memory-bound kernel — FMA
  // contributes nothing here anyway)
#pragma clang fp contract(off)
  // per-chunk quant params staged through LDS once per block
  __shared__ float s_lower[DQR_MAX_CHUNKS];
  __shared__ float s_inv_scale[DQR_MAX_CHUNKS];
  for (int c = threadIdx.x; c < num_chunks; c += blockDim.x) {
    const float* hdr =
        reinterpret_cast<const float*>(in + (size_t)c * chunk_stride);
    float scale, lower, upper;
    quant_params(hdr[0], hdr[1], scale, lower, upper);
    s_lower[c] = lower;
    s_inv_scale[c] = 1.0f / scale;
  }
  __syncthreads();

  // per-dtype width, same tradeoff as dequantize_kernel (measured
  // r2c9): f32 V=4, f16/bf16 V=16
  constexpr int V = sizeof(T) == 4 ? 4 : 16;
  constexpr int OV = Vec16<T>::N;       // elems per 16B output vector
  constexpr int NOUT = V / OV;          // output vectors per iteration
  using VT = Vec16<T>;
  const size_t tid = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const size_t nv = chunk / V;
  VT* dv = reinterpret_cast<VT*>(dst);
  for (size_t i = tid; i < nv; i += stride) {
    float acc[V];
#pragma unroll
    for (int k = 0; k < V; ++k) acc[k] = 0.f;
    for (int c = 0; c < num_chunks; ++c) {
      const uint8_t* payload = in + (size_t)c * chunk_stride + 32;
      alignas(16) uint8_t q[V];
      if (V == 4)
        *reinterpret_cast<uint32_t*>(q) =
            *reinterpret_cast<const uint32_t*>(payload + i * 4);
      else
        *reinterpret_cast<uint4*>(q) =
            *reinterpret_cast<const uint4*>(payload + i * 16);
      const float lower = s_lower[c], inv_scale = s_inv_scale[c];
#pragma unroll
      for (int k = 0; k < V; ++k)
        acc[k] += to_f(from_f<T>(((float)q[k] + lower) * inv_scale));
    }
#pragma unroll
    for (int o = 0; o < NOUT; ++o) {
      VT out;
#pragma unroll
      for (int k = 0; k < OV; ++k)
        out.v[k] = from_f<T>(acc[o * OV + k] * post_scale);
      dv[i * NOUT + o] = out;
    }
  }
  for (size_t i = nv * V + tid; i < chunk; i += stride) {
    float acc = 0.f;
    for (int c = 0; c < num_chunks; ++c) {
      const uint8_t* payload = in + (size_t)c * chunk_stride + 32;
      acc += to_f(from_f<T>(((float)payload[i] + s_lower[c])
                            * s_inv_scale[c]));
    }
    dst[i] = from_f<T>(acc * post_scale);
  }
}

// ---------------------------------------------------------------------------
// host-side launchers (extern "C", stream-passing; no syncs)
// ---------------------------------------------------------------------------

static inline int grid_for(size_t work_items) {
  size_t g = (work_items + BLOCK - 1) / BLOCK;
  if (g > MAX_GRID) g = MAX_GRID;
  if (g < 1) g = 1;
  return (int)g;
}

template <typename T, int OP>
static void launch_ew(void* x, const void* y, float f, size_t n,
                      hipStream_t stream) {
  int grid = grid_for(n / (2 * Vec16<T>::N) + 1);
  hipLaunchKernelGGL((ew_kernel<T, OP>), dim3(grid), dim3(BLOCK), 0, stream,
                     (T*)x, (const T*)y, f, n);
}

extern "C" {

// dtype: 0=f32, 1=f16, 2=bf16
void bagua_ew_launch(int op, int dtype, void* x, const void* y, float f,
                     size_t n, hipStream_t stream) {
  if (op == EW_AVERAGE) {
    switch (dtype) {
      case 0: launch_ew<float, EW_AVERAGE>(x, y, f, n, stream); break;
      case 1: launch_ew<__half, EW_AVERAGE>(x, y, f, n, stream); break;
      case 2: launch_ew<__hip_bfloat16, EW_AVERAGE>(x, y, f, n, stream); break;
    }
  } else if (op == EW_ADD) {
    switch (dtype) {
      case 0: launch_ew<float, EW_ADD>(x, y, f, n, stream); break;
      case 1: launch_ew<__half, EW_ADD>(x, y, f, n, stream); break;
      case 2: launch_ew<__hip_bfloat16, EW_ADD>(x, y, f, n, stream); break;
    }
  } else if (op == EW_SUB) {
    switch (dtype) {
      case 0: launch_ew<float, EW_SUB>(x, y, f, n, stream); break;
      case 1: launch_ew<__half, EW_SUB>(x, y, f, n, stream); break;
      case 2: launch_ew<__hip_bfloat16, EW_SUB>(x, y, f, n, stream); break;
    }
  } else if (op == EW_ADDMUL) {
    switch (dtype) {
      case 0: launch_ew<float, EW_ADDMUL>(x, y, f, n, stream); break;
      case 1: launch_ew<__half, EW_ADDMUL>(x, y, f, n, stream); break;
      case 2: launch_ew<__hip_bfloat16, EW_ADDMUL>(x, y, f, n, stream); break;
    }
  } else if (op == EW_DIV) {
    switch (dtype) {
      case 0: launch_ew<float, EW_DIV>(x, y, f, n, stream); break;
      case 1: launch_ew<__half, EW_DIV>(x, y, f, n, stream); break;
      case 2: launch_ew<__hip_bfloat16, EW_DIV>(x, y, f, n, stream); break;
    }
  }
}

void bagua_async_avg_launch(int dtype, void* x, const void* reduced,
                            const void* x_copy, float nranks, size_t n,
                            hipStream_t stream) {
  int grid = grid_for(n / 4 + 1);
  float inv_n = 1.0f / nranks;
  switch (dtype) {
    case 0:
      hipLaunchKernelGGL((async_avg_kernel<float>), dim3(grid), dim3(BLOCK),
                         0, stream, (float*)x, (const float*)reduced,
                         (const float*)x_copy, inv_n, n);
      break;
    case 1:
      hipLaunchKernelGGL((async_avg_kernel<__half>), dim3(grid), dim3(BLOCK),
                         0, stream, (__half*)x, (const __half*)reduced,
                         (const __half*)x_copy, inv_n, n);
      break;
    case 2:
      hipLaunchKernelGGL((async_avg_kernel<__hip_bfloat16>), dim3(grid),
                         dim3(BLOCK), 0, stream, (__hip_bfloat16*)x,
                         (const __hip_bfloat16*)reduced,
                         (const __hip_bfloat16*)x_copy, inv_n, n);
      break;
  }
}

void bagua_reduce_chunk_launch(int dtype, void* x, int num_chunks,
                               int target_chunk, int average, size_t chunk,
                               hipStream_t stream) {
  int grid = grid_for(chunk / 4 + 1);
  float post = average ? 1.0f / num_chunks : 1.0f;
  switch (dtype) {
    case 0:
      hipLaunchKernelGGL((reduce_chunk_kernel<float>), dim3(grid),
                         dim3(BLOCK), 0, stream, (float*)x, num_chunks,
                         target_chunk, post, chunk);
      break;
    case 1:
      hipLaunchKernelGGL((reduce_chunk_kernel<__half>), dim3(grid),
                         dim3(BLOCK), 0, stream, (__half*)x, num_chunks,
                         target_chunk, post, chunk);
      break;
    case 2:
      hipLaunchKernelGGL((reduce_chunk_kernel<__hip_bfloat16>), dim3(grid),
                         dim3(BLOCK), 0, stream, (__hip_bfloat16*)x,
                         num_chunks, target_chunk, post, chunk);
      break;
  }
}

// scratch: uint32[2*num_chunks_total]; out: chunked wire buffer
void bagua_compress_launch(int dtype, const void* x, uint8_t* out,
                           uint32_t* scratch, uint32_t* partials,
                           size_t chunk, size_t chunk_stride,
                           int num_chunks_total, int chunk_begin,
                           int chunk_count, hipStream_t stream) {
  int gx = grid_for(chunk / 8 + 1);
  if (gx > MINMAX_MAX_BLOCKS) gx = MINMAX_MAX_BLOCKS;
  dim3 grid(gx, chunk_count);
  switch (dtype) {
    case 0:
      hipLaunchKernelGGL((minmax_kernel<float>), grid, dim3(BLOCK), 0,
                         stream, (const float*)x, chunk, chunk_begin,
                         partials);
      hipLaunchKernelGGL(minmax_finalize_kernel, dim3(chunk_count),
                         dim3(BLOCK), 0, stream, partials, gx,
                         chunk_begin, scratch);
      hipLaunchKernelGGL((quantize_kernel<float>), grid, dim3(BLOCK), 0,
                         stream, (const float*)x, chunk, chunk_begin,
                         chunk_stride, scratch, out);
      break;
    case 1:
      hipLaunchKernelGGL((minmax_kernel<__half>), grid, dim3(BLOCK), 0,
                         stream, (const __half*)x, chunk, chunk_begin,
                         partials);
      hipLaunchKernelGGL(minmax_finalize_kernel, dim3(chunk_count),
                         dim3(BLOCK), 0, stream, partials, gx,
                         chunk_begin, scratch);
      hipLaunchKernelGGL((quantize_kernel<__half>), grid, dim3(BLOCK), 0,
                         stream, (const __half*)x, chunk, chunk_begin,
                         chunk_stride, scratch, out);
      break;
    case 2:
      hipLaunchKernelGGL((minmax_kernel<__hip_bfloat16>), grid, dim3(BLOCK),
                         0, stream, (const __hip_bfloat16*)x, chunk,
                         chunk_begin, partials);
      hipLaunchKernelGGL(minmax_finalize_kernel, dim3(chunk_count),
                         dim3(BLOCK), 0, stream, partials, gx,
                         chunk_begin, scratch);
      hipLaunchKernelGGL((quantize_kernel<__hip_bfloat16>), grid,
                         dim3(BLOCK), 0, stream, (const __hip_bfloat16*)x,
                         chunk, chunk_begin, chunk_stride, scratch, out);
      break;
  }
}

void bagua_decompress_launch(int dtype, const uint8_t* in, void* x,
                             size_t chunk, size_t chunk_stride,
                             int chunk_begin, int chunk_count,
                             hipStream_t stream) {
  int gx = grid_for(chunk / 8 + 1);
  if (gx > 512) gx = 512;
  dim3 grid(gx, chunk_count);
  switch (dtype) {
    case 0:
      hipLaunchKernelGGL((dequantize_kernel<float>), grid, dim3(BLOCK), 0,
                         stream, in, chunk, chunk_begin, chunk_stride,
                         (float*)x);
      break;
    case 1:
      hipLaunchKernelGGL((dequantize_kernel<__half>), grid, dim3(BLOCK), 0,
                         stream, in, chunk, chunk_begin, chunk_stride,
                         (__half*)x);
      break;
    case 2:
      hipLaunchKernelGGL((dequantize_kernel<__hip_bfloat16>), grid,
                         dim3(BLOCK), 0, stream, in, chunk, chunk_begin,
                         chunk_stride, (__hip_bfloat16*)x);
      break;
  }
}

// fused dequantize + reduce of all wire chunks into flat's target chunk
// (x points at the BUCKET base; target offset applied here)
void bagua_dequant_reduce_launch(int dtype, const uint8_t* in, void* x,
                                 size_t chunk, size_t chunk_stride,
                                 int num_chunks, int target_chunk,
                                 int average, hipStream_t stream) {
  float post_scale = average ? 1.0f / (float)num_chunks : 1.0f;
  int grid = grid_for(chunk / 16 + 1);
  switch (dtype) {
    case 0:
      hipLaunchKernelGGL((dequant_reduce_kernel<float>), dim3(grid),
                         dim3(BLOCK), 0, stream, in, chunk, chunk_stride,
                         num_chunks, post_scale,
                         (float*)x + (size_t)target_chunk * chunk);
      break;
    case 1:
      hipLaunchKernelGGL((dequant_reduce_kernel<__half>), dim3(grid),
                         dim3(BLOCK), 0, stream, in, chunk, chunk_stride,
                         num_chunks, post_scale,
                         (__half*)x + (size_t)target_chunk * chunk);
      break;
    case 2:
      hipLaunchKernelGGL((dequant_reduce_kernel<__hip_bfloat16>),
                         dim3(grid), dim3(BLOCK), 0, stream, in, chunk,
                         chunk_stride, num_chunks, post_scale,
                         (__hip_bfloat16*)x + (size_t)target_chunk * chunk);
      break;
  }
}

}  // extern "C"

// ---------------------------------------------------------------------------
// fused optimizer steps (f32 master weights) — one pass over the flat
// param group: grad transform + momentum/variance update + weight update
// in a single memory-bound kernel instead of the 3-5 launches of a
// generic foreach optimizer. MI355X-native addition (the reference fused
// by contiguity only; north-star asks for the fused step as a HIP
// kernel).
// ---------------------------------------------------------------------------

extern "C" {

// SGD with momentum/dampening/nesterov/weight-decay (torch semantics)
__global__ void fused_sgd_kernel(float* __restrict__ p,
                                 const float* __restrict__ g,
                                 float* __restrict__ m, float lr,
                                 float momentum, float dampening,
                                 float weight_decay, int nesterov,
                                 int momentum_initialized, size_t n) {
  const size_t tid = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  // two float4 per lane per iteration (32 B accesses, +5-9% on the RMW
  // mix per the elementwise measurements)
  const size_t nv = n / 8;
  float4* p4 = reinterpret_cast<float4*>(p);
  const float4* g4 = reinterpret_cast<const float4*>(g);
  float4* m4 = reinterpret_cast<float4*>(m);
  for (size_t i = tid; i < nv; i += stride) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const size_t idx = 2 * i + j;
      float4 pv = p4[idx], gv = g4[idx];
      float4 mv = momentum != 0.f ? m4[idx] : make_float4(0, 0, 0, 0);
      float* pp = &pv.x;
      const float* gg = &gv.x;
      float* mm = &mv.x;
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        float grad = gg[k] + weight_decay * pp[k];
        float upd = grad;
        if (momentum != 0.f) {
          float buf = momentum_initialized
                          ? momentum * mm[k] + (1.f - dampening) * grad
                          : grad;
          mm[k] = buf;
          upd = nesterov ? grad + momentum * buf : buf;
        }
        pp[k] -= lr * upd;
      }
      p4[idx] = pv;
      if (momentum != 0.f) m4[idx] = mv;
    }
  }
  for (size_t i = nv * 8 + tid; i < n; i += stride) {
    float grad = g[i] + weight_decay * p[i];
    float upd = grad;
    if (momentum != 0.f) {
      float buf = momentum_initialized
                      ? momentum * m[i] + (1.f - dampening) * grad
                      : grad;
      m[i] = buf;
      upd = nesterov ? grad + momentum * buf : buf;
    }
    p[i] -= lr * upd;
  }
}

// Adam / AdamW (torch semantics, bias-corrected)
__global__ void fused_adam_kernel(float* __restrict__ p,
                                  const float* __restrict__ g,
                                  float* __restrict__ m,
                                  float* __restrict__ v, float lr,
                                  float beta1, float beta2, float eps,
                                  float weight_decay, int adamw,
                                  float bias_correction1,
                                  float bias_correction2, size_t n) {
  const size_t tid = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const float inv_bc1 = 1.f / bias_correction1;
  const float inv_sqrt_bc2 = rsqrtf(bias_correction2);
  // vectorized: two float4 per lane per array (32 B accesses; the
  // scalar form measured 5.0 TB/s vs the ~7 TB/s RMW ceiling)
  const size_t nv = n / 8;
  float4* p4 = reinterpret_cast<float4*>(p);
  const float4* g4 = reinterpret_cast<const float4*>(g);
  float4* m4 = reinterpret_cast<float4*>(m);
  float4* v4 = reinterpret_cast<float4*>(v);
  for (size_t i = tid; i < nv; i += stride) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const size_t idx = 2 * i + j;
      float4 pv = p4[idx], gv = g4[idx], mv = m4[idx], vv = v4[idx];
      float* pp = &pv.x;
      float* gg = &gv.x;
      float* mm = &mv.x;
      float* uu = &vv.x;
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        float param = pp[k];
        float grad = gg[k];
        if (adamw)
          param *= (1.f - lr * weight_decay);
        else
          grad += weight_decay * param;
        float mi = beta1 * mm[k] + (1.f - beta1) * grad;
        float vi = beta2 * uu[k] + (1.f - beta2) * grad * grad;
        mm[k] = mi;
        uu[k] = vi;
        float denom = sqrtf(vi) * inv_sqrt_bc2 + eps;
        pp[k] = param - lr * inv_bc1 * mi / denom;
      }
      p4[idx] = pv;
      m4[idx] = mv;
      v4[idx] = vv;
    }
  }
  for (size_t i = nv * 8 + tid; i < n; i += stride) {
    float param = p[i];
    float grad = g[i];
    if (adamw)
      param *= (1.f - lr * weight_decay);
    else
      grad += weight_decay * param;
    float mi = beta1 * m[i] + (1.f - beta1) * grad;
    float vi = beta2 * v[i] + (1.f - beta2) * grad * grad;
    m[i] = mi;
    v[i] = vi;
    float denom = sqrtf(vi) * inv_sqrt_bc2 + eps;
    p[i] = param - lr * inv_bc1 * mi / denom;
  }
}

void bagua_fused_sgd_launch(float* p, const float* g, float* m, float lr,
                            float momentum, float dampening,
                            float weight_decay, int nesterov,
                            int momentum_initialized, size_t n,
                            hipStream_t stream) {
  int grid = grid_for(n / 8 + 1);
  hipLaunchKernelGGL(fused_sgd_kernel, dim3(grid), dim3(BLOCK), 0, stream,
                     p, g, m, lr, momentum, dampening, weight_decay,
                     nesterov, momentum_initialized, n);
}


// bf16 weights + fp32 master + bf16 grads: the pure-bf16 training path
// (fp32 master keeps SGD exact; bf16 params keep fwd/bwd and the grad
// allreduce at half the bytes)
__global__ void fused_sgd_mixed_kernel(__hip_bfloat16* __restrict__ p,
                                       const __hip_bfloat16* __restrict__ g,
                                       float* __restrict__ master,
                                       float* __restrict__ m, float lr,
                                       float momentum, float dampening,
                                       float weight_decay, int nesterov,
                                       int momentum_initialized, size_t n) {
  const size_t tid = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  // 8 elements per lane: one 16 B bf16 load/store + two float4 master/
  // momentum accesses (the scalar form paid the 2-2.5x slow bf16
  // scalar-load path on the flagship pure-bf16 bench)
  const size_t nv = n / 8;
  using BV = Vec16<__hip_bfloat16>;  // 8 bf16
  BV* p8 = reinterpret_cast<BV*>(p);
  const BV* g8 = reinterpret_cast<const BV*>(g);
  float4* ma4 = reinterpret_cast<float4*>(master);
  float4* m4 = reinterpret_cast<float4*>(m);
  for (size_t i = tid; i < nv; i += stride) {
    BV gv = g8[i];
    BV pv;
    float4 w[2] = {ma4[2 * i], ma4[2 * i + 1]};
    float4 mb[2];
    if (momentum != 0.f && momentum_initialized) {
      mb[0] = m4[2 * i];
      mb[1] = m4[2 * i + 1];
    }
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float* ww = &w[k / 4].x;
      float* mm = &mb[k / 4].x;
      float wk = ww[k % 4];
      float grad = __bfloat162float(gv.v[k]) + weight_decay * wk;
      float upd = grad;
      if (momentum != 0.f) {
        float buf = momentum_initialized
                        ? momentum * mm[k % 4] + (1.f - dampening) * grad
                        : grad;
        mm[k % 4] = buf;
        upd = nesterov ? grad + momentum * buf : buf;
      }
      wk -= lr * upd;
      ww[k % 4] = wk;
      pv.v[k] = __float2bfloat16(wk);
    }
    ma4[2 * i] = w[0];
    ma4[2 * i + 1] = w[1];
    if (momentum != 0.f) {
      m4[2 * i] = mb[0];
      m4[2 * i + 1] = mb[1];
    }
    p8[i] = pv;
  }
  for (size_t i = nv * 8 + tid; i < n; i += stride) {
    float w = master[i];
    float grad = __bfloat162float(g[i]) + weight_decay * w;
    float upd = grad;
    if (momentum != 0.f) {
      float buf = momentum_initialized
                      ? momentum * m[i] + (1.f - dampening) * grad
                      : grad;
      m[i] = buf;
      upd = nesterov ? grad + momentum * buf : buf;
    }
    w -= lr * upd;
    master[i] = w;
    p[i] = __float2bfloat16(w);
  }
}

void bagua_fused_sgd_mixed_launch(void* p, const void* g, float* master,
                                  float* m, float lr, float momentum,
                                  float dampening, float weight_decay,
                                  int nesterov, int momentum_initialized,
                                  size_t n, hipStream_t stream) {
  int grid = grid_for(n / 8 + 1);
  hipLaunchKernelGGL(fused_sgd_mixed_kernel, dim3(grid), dim3(BLOCK), 0,
                     stream, (__hip_bfloat16*)p, (const __hip_bfloat16*)g,
                     master, m, lr, momentum, dampening, weight_decay,
                     nesterov, momentum_initialized, n);
}

void bagua_fused_adam_launch(float* p, const float* g, float* m, float* v,
                             float lr, float beta1, float beta2, float eps,
                             float weight_decay, int adamw, float bc1,
                             float bc2, size_t n, hipStream_t stream) {
  int grid = grid_for(n / 8 + 1);
  hipLaunchKernelGGL(fused_adam_kernel, dim3(grid), dim3(BLOCK), 0, stream,
                     p, g, m, v, lr, beta1, beta2, eps, weight_decay,
                     adamw, bc1, bc2, n);
}

}  // extern "C"

// ---------------------------------------------------------------------------
// p2p alltoall support: cross-GPU flag barrier over xGMI.
//
// The 8-GPU MI355X node is FULLY CONNECTED point-to-point (7 links/GPU),
// so a one-hop alltoall is: every rank pulls its chunk from each peer's
// exported send buffer. Ordering needs a device-side barrier — IPC
// events re-recorded every round race (a peer's wait can capture the
// previous recording), so ranks rendezvous on a monotonically increasing
// sequence number in FINE-GRAINED peer-visible memory instead:
//   arrive: write seq into MY slot of every peer's flag array
//   wait:   spin until every slot of MY flag array reaches seq
// One wave handles all peers (nranks <= 64 by construction).
// ---------------------------------------------------------------------------

__global__ void p2p_barrier_kernel(
    unsigned long long* const* __restrict__ peer_flags,  // [nranks] ptrs
    volatile unsigned long long* __restrict__ my_flags,  // [nranks]
    int rank, int nranks, unsigned long long seq) {
  const int p = threadIdx.x;
  if (p >= nranks) return;
  // make every prior write to my send buffer visible system-wide before
  // announcing arrival
  __threadfence_system();
  volatile unsigned long long* slot =
      (volatile unsigned long long*)(peer_flags[p]) + rank;
  *slot = seq;
  __threadfence_system();
  while (my_flags[p] < seq) {
    __builtin_amdgcn_s_sleep(8);
  }
  __threadfence_system();
}

extern "C" void bagua_p2p_barrier_launch(void* peer_flags_dev_array,
                                         void* my_flags, int rank,
                                         int nranks,
                                         unsigned long long seq,
                                         hipStream_t stream) {
  hipLaunchKernelGGL(p2p_barrier_kernel, dim3(1), dim3(64), 0, stream,
                     (unsigned long long* const*)peer_flags_dev_array,
                     (volatile unsigned long long*)my_flags, rank, nranks,
                     seq);
}
