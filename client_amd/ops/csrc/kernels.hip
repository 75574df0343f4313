// client_amd CDNA4 kernels — single compilation unit shared by the
// Python extension (_hip_c) and the C++ client library. All launchers
// are C ABI taking an explicit hipStream_t so both stacks compose the
// kernels with their own streams/graphs. See client_amd/ops/csrc
// provenance comments on each kernel.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>

#include <cstdint>

namespace {

inline int ca_grid_for(long work_items) {
  // memory-bound streaming grid: cap at 2048 workgroups, grid-stride
  // the rest (256 CUs x 8 blocks/CU)
  long blocks = (work_items + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

}  // namespace

// ---------------------------------------------------------------------------
// Kernels
// ---------------------------------------------------------------------------

// fp32 -> bf16 by truncation (keep high 16 bits) — byte-exact with the
// wire codec in client_amd.utils.serialize_bf16_tensor (reference
// semantics: utils/__init__.py:294-330). Vectorized 8 elems/lane:
// 32 B in, 16 B out per lane.
extern "C" __global__ void cast_fp32_bf16_kernel(const uint32_t* __restrict__ src,
                                                 uint16_t* __restrict__ dst,
                                                 long n) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i + 8 <= n; i += stride) {
    const uint4 a = *reinterpret_cast<const uint4*>(src + i);
    const uint4 b = *reinterpret_cast<const uint4*>(src + i + 4);
    uint16_t out[8];
    out[0] = (uint16_t)(a.x >> 16);
    out[1] = (uint16_t)(a.y >> 16);
    out[2] = (uint16_t)(a.z >> 16);
    out[3] = (uint16_t)(a.w >> 16);
    out[4] = (uint16_t)(b.x >> 16);
    out[5] = (uint16_t)(b.y >> 16);
    out[6] = (uint16_t)(b.z >> 16);
    out[7] = (uint16_t)(b.w >> 16);
    *reinterpret_cast<uint4*>(dst + i) = *reinterpret_cast<uint4*>(out);
  }
  // tail (grid-stride tail handling: only the lanes owning the ragged end)
  long tail_start = (n / 8) * 8;
  long ti = tail_start + (blockIdx.x * blockDim.x + threadIdx.x);
  if (ti < n && (blockIdx.x * blockDim.x + threadIdx.x) < 8) {
    dst[ti] = (uint16_t)(src[ti] >> 16);
  }
}

// bf16 -> fp32 by zero-extension (wire-exact inverse).
extern "C" __global__ void cast_bf16_fp32_kernel(const uint16_t* __restrict__ src,
                                                 uint32_t* __restrict__ dst,
                                                 long n) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i + 8 <= n; i += stride) {
    const uint4 a = *reinterpret_cast<const uint4*>(src + i);  // 8 bf16
    const uint16_t* s = reinterpret_cast<const uint16_t*>(&a);
    uint4 lo, hi;
    lo.x = (uint32_t)s[0] << 16;
    lo.y = (uint32_t)s[1] << 16;
    lo.z = (uint32_t)s[2] << 16;
    lo.w = (uint32_t)s[3] << 16;
    hi.x = (uint32_t)s[4] << 16;
    hi.y = (uint32_t)s[5] << 16;
    hi.z = (uint32_t)s[6] << 16;
    hi.w = (uint32_t)s[7] << 16;
    *reinterpret_cast<uint4*>(dst + i) = lo;
    *reinterpret_cast<uint4*>(dst + i + 4) = hi;
  }
  long tail_start = (n / 8) * 8;
  long ti = tail_start + (blockIdx.x * blockDim.x + threadIdx.x);
  if (ti < n && (blockIdx.x * blockDim.x + threadIdx.x) < 8) {
    dst[ti] = (uint32_t)src[ti] << 16;
  }
}

// 16-elem/lane unpack variant (mirrors the pack-kernel A/B win)
extern "C" __global__ void cast_bf16_fp32_v2_kernel(
    const uint16_t* __restrict__ src, uint32_t* __restrict__ dst, long n) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 16;
  long stride = (long)gridDim.x * blockDim.x * 16;
  for (long i = i0; i + 16 <= n; i += stride) {
    uint4 a[2];
    a[0] = *reinterpret_cast<const uint4*>(src + i);
    a[1] = *reinterpret_cast<const uint4*>(src + i + 8);
    const uint16_t* e = reinterpret_cast<const uint16_t*>(a);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      uint4 o;
      o.x = (uint32_t)e[4 * j + 0] << 16;
      o.y = (uint32_t)e[4 * j + 1] << 16;
      o.z = (uint32_t)e[4 * j + 2] << 16;
      o.w = (uint32_t)e[4 * j + 3] << 16;
      *reinterpret_cast<uint4*>(dst + i + 4 * j) = o;
    }
  }
  long tail_start = (n / 16) * 16;
  long ti = tail_start + (blockIdx.x * blockDim.x + threadIdx.x);
  if (ti < n && (blockIdx.x * blockDim.x + threadIdx.x) < 16) {
    dst[ti] = (uint32_t)src[ti] << 16;
  }
}

// fp32 -> fp8 e4m3 (OCP fn, the CDNA4-native format — NOT MI300X fnuz;
// cdna_hip_programming.md §4). RNE via the __hip_fp8_e4m3 HW convert.
extern "C" __global__ void cast_fp32_fp8e4m3_kernel(const float* __restrict__ src,
                                                    uint8_t* __restrict__ dst,
                                                    long n) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i + 8 <= n; i += stride) {
    const float4 a = *reinterpret_cast<const float4*>(src + i);
    const float4 b = *reinterpret_cast<const float4*>(src + i + 4);
    uint8_t out[8];
    out[0] = __hip_fp8_e4m3(a.x).__x;
    out[1] = __hip_fp8_e4m3(a.y).__x;
    out[2] = __hip_fp8_e4m3(a.z).__x;
    out[3] = __hip_fp8_e4m3(a.w).__x;
    out[4] = __hip_fp8_e4m3(b.x).__x;
    out[5] = __hip_fp8_e4m3(b.y).__x;
    out[6] = __hip_fp8_e4m3(b.z).__x;
    out[7] = __hip_fp8_e4m3(b.w).__x;
    *reinterpret_cast<uint64_t*>(dst + i) = *reinterpret_cast<uint64_t*>(out);
  }
  long tail_start = (n / 8) * 8;
  long ti = tail_start + (blockIdx.x * blockDim.x + threadIdx.x);
  if (ti < n && (blockIdx.x * blockDim.x + threadIdx.x) < 8) {
    dst[ti] = __hip_fp8_e4m3(src[ti]).__x;
  }
}

extern "C" __global__ void cast_fp8e4m3_fp32_kernel(const uint8_t* __restrict__ src,
                                                    float* __restrict__ dst,
                                                    long n) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i + 8 <= n; i += stride) {
    uint64_t packed = *reinterpret_cast<const uint64_t*>(src + i);
    const uint8_t* s = reinterpret_cast<const uint8_t*>(&packed);
    float4 lo, hi;
    __hip_fp8_e4m3 v;
    v.__x = s[0]; lo.x = float(v);
    v.__x = s[1]; lo.y = float(v);
    v.__x = s[2]; lo.z = float(v);
    v.__x = s[3]; lo.w = float(v);
    v.__x = s[4]; hi.x = float(v);
    v.__x = s[5]; hi.y = float(v);
    v.__x = s[6]; hi.z = float(v);
    v.__x = s[7]; hi.w = float(v);
    *reinterpret_cast<float4*>(dst + i) = lo;
    *reinterpret_cast<float4*>(dst + i + 4) = hi;
  }
  long tail_start = (n / 8) * 8;
  long ti = tail_start + (blockIdx.x * blockDim.x + threadIdx.x);
  if (ti < n && (blockIdx.x * blockDim.x + threadIdx.x) < 8) {
    __hip_fp8_e4m3 v;
    v.__x = src[ti];
    dst[ti] = float(v);
  }
}

// A/B variant: 16 elems/lane (64 B loads, 32 B stores per lane per
// iteration) — measured against the 8-elem kernel on hardware.
extern "C" __global__ void cast_fp32_bf16_v2_kernel(
    const uint32_t* __restrict__ src, uint16_t* __restrict__ dst, long n) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 16;
  long stride = (long)gridDim.x * blockDim.x * 16;
  for (long i = i0; i + 16 <= n; i += stride) {
    uint4 a[4];
#pragma unroll
    for (int j = 0; j < 4; ++j)
      a[j] = *reinterpret_cast<const uint4*>(src + i + 4 * j);
    uint16_t out[16];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      out[4 * j + 0] = (uint16_t)(a[j].x >> 16);
      out[4 * j + 1] = (uint16_t)(a[j].y >> 16);
      out[4 * j + 2] = (uint16_t)(a[j].z >> 16);
      out[4 * j + 3] = (uint16_t)(a[j].w >> 16);
    }
    *reinterpret_cast<uint4*>(dst + i) = *reinterpret_cast<uint4*>(out);
    *reinterpret_cast<uint4*>(dst + i + 8) =
        *reinterpret_cast<uint4*>(out + 8);
  }
  long tail_start = (n / 16) * 16;
  long ti = tail_start + (blockIdx.x * blockDim.x + threadIdx.x);
  if (ti < n && (blockIdx.x * blockDim.x + threadIdx.x) < 16) {
    dst[ti] = (uint16_t)(src[ti] >> 16);
  }
}

// A/B variant 3: 16 elems/lane with nontemporal loads/stores (the
// packed stream is written once and read by a different consumer —
// bypassing L2 may help at HBM-bound sizes).
extern "C" __global__ void cast_fp32_bf16_v3_kernel(
    const uint32_t* __restrict__ src, uint16_t* __restrict__ dst, long n) {
  // clang's nontemporal builtins need ext_vector types, not the HIP
  // vector structs
  typedef unsigned int u32x4 __attribute__((ext_vector_type(4)));
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 16;
  long stride = (long)gridDim.x * blockDim.x * 16;
  for (long i = i0; i + 16 <= n; i += stride) {
    u32x4 a[4];
#pragma unroll
    for (int j = 0; j < 4; ++j)
      a[j] = __builtin_nontemporal_load(
          reinterpret_cast<const u32x4*>(src + i + 4 * j));
    uint16_t out[16];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      out[4 * j + 0] = (uint16_t)(a[j].x >> 16);
      out[4 * j + 1] = (uint16_t)(a[j].y >> 16);
      out[4 * j + 2] = (uint16_t)(a[j].z >> 16);
      out[4 * j + 3] = (uint16_t)(a[j].w >> 16);
    }
    __builtin_nontemporal_store(*reinterpret_cast<u32x4*>(out),
                                reinterpret_cast<u32x4*>(dst + i));
    __builtin_nontemporal_store(*reinterpret_cast<u32x4*>(out + 8),
                                reinterpret_cast<u32x4*>(dst + i + 8));
  }
  long tail_start = (n / 16) * 16;
  long ti = tail_start + (blockIdx.x * blockDim.x + threadIdx.x);
  if (ti < n && (blockIdx.x * blockDim.x + threadIdx.x) < 16) {
    dst[ti] = (uint16_t)(src[ti] >> 16);
  }
}

// Fused RMSNorm for bf16 rows: out = x * rsqrt(mean(x^2)+eps) * w,
// computed in fp32 (byte-compatible with the torch reference sequence
// float() -> pow/mean/rsqrt -> mul -> to(bf16), which launches ~7
// kernels; this is one). One workgroup per row; dim must be a multiple
// of 8 for the vectorized loads (4096/1024/... in practice).
extern "C" __global__ void rmsnorm_bf16_kernel(
    const uint16_t* __restrict__ x, const uint16_t* __restrict__ w,
    uint16_t* __restrict__ out, int dim, float eps) {
  const int row = blockIdx.x;
  const uint16_t* xr = x + (long)row * dim;
  uint16_t* outr = out + (long)row * dim;
  float acc = 0.f;
  for (int i = threadIdx.x * 8; i < dim; i += blockDim.x * 8) {
    uint4 v = *reinterpret_cast<const uint4*>(xr + i);
    const uint16_t* e = reinterpret_cast<const uint16_t*>(&v);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = __uint_as_float((uint32_t)e[j] << 16);
      acc += f * f;
    }
  }
  // wave + LDS reduction (64-wide waves)
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off);
  __shared__ float warp_sums[16];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  if (lane == 0) warp_sums[wave] = acc;
  __syncthreads();
  const int n_waves = (blockDim.x + 63) >> 6;
  if (threadIdx.x == 0) {
    float total = 0.f;
    for (int i = 0; i < n_waves; ++i) total += warp_sums[i];
    warp_sums[0] = rsqrtf(total / dim + eps);
  }
  __syncthreads();
  const float scale = warp_sums[0];
  for (int i = threadIdx.x * 8; i < dim; i += blockDim.x * 8) {
    uint4 v = *reinterpret_cast<const uint4*>(xr + i);
    uint4 wv = *reinterpret_cast<const uint4*>(w + i);
    const uint16_t* e = reinterpret_cast<const uint16_t*>(&v);
    const uint16_t* we = reinterpret_cast<const uint16_t*>(&wv);
    uint16_t o[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = __uint_as_float((uint32_t)e[j] << 16);
      float wf = __uint_as_float((uint32_t)we[j] << 16);
      o[j] = __bfloat16_as_ushort(__float2bfloat16(f * scale * wf));
    }
    *reinterpret_cast<uint4*>(outr + i) = *reinterpret_cast<uint4*>(o);
  }
}

// Fused decode-step RoPE for bf16 q AND k in one launch, with per-row
// positions (continuous batching): q [b, hq, d], k [b, hk, d]
// contiguous (s=1), cos/sin tables [max_seq, d/2] fp32, pos [b] int64.
// Replaces ~8 slicing/elementwise launches per projection.
extern "C" __global__ void rope_decode_bf16_kernel(
    uint16_t* __restrict__ q, uint16_t* __restrict__ k,
    const float* __restrict__ cos_tab, const float* __restrict__ sin_tab,
    const long* __restrict__ pos, int b, int hq, int hk, int d) {
  const int half = d / 2;
  const long total = (long)b * (hq + hk) * half;
  long i = (long)(blockIdx.x * blockDim.x + threadIdx.x);
  const long stride = (long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    long rem = i;
    const int j = (int)(rem % half);  // rotation pair index
    rem /= half;
    const int head = (int)(rem % (hq + hk));
    const int row = (int)(rem / (hq + hk));
    uint16_t* base = (head < hq)
                         ? q + ((long)row * hq + head) * d
                         : k + ((long)row * hk + (head - hq)) * d;
    const float c = cos_tab[pos[row] * half + j];
    const float s = sin_tab[pos[row] * half + j];
    const float x1 = __uint_as_float((uint32_t)base[2 * j] << 16);
    const float x2 = __uint_as_float((uint32_t)base[2 * j + 1] << 16);
    base[2 * j] = __bfloat16_as_ushort(__float2bfloat16(x1 * c - x2 * s));
    base[2 * j + 1] = __bfloat16_as_ushort(__float2bfloat16(x1 * s + x2 * c));
  }
}

// RoPE + KV-cache scatter in ONE launch: rotates q (in place) and k
// (rotated values written straight into the cache at each row's
// position), and copies v into the cache — removing the two
// torch index_put launches per layer per decode step (64/step at 32
// layers; measured 28.9 ms / 5120 calls in the decode profile).
// ck/cv: [b, hk, cache_len, d]; k/v: [b, hk, d] contiguous.
extern "C" __global__ void rope_scatter_decode_bf16_kernel(
    uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, uint16_t* __restrict__ ck,
    uint16_t* __restrict__ cv, const float* __restrict__ cos_tab,
    const float* __restrict__ sin_tab, const long* __restrict__ pos, int b,
    int hq, int hk, int d, long cache_len, float q_scale) {
  const int half = d / 2;
  const long total = (long)b * (hq + 2 * hk) * half;
  long i = (long)(blockIdx.x * blockDim.x + threadIdx.x);
  const long stride = (long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    long rem = i;
    const int j = (int)(rem % half);
    rem /= half;
    const int head = (int)(rem % (hq + 2 * hk));
    const int row = (int)(rem / (hq + 2 * hk));
    const long p = pos[row];
    if (head < hq) {  // q: rotate in place (+ folded 1/sqrt(d) scale)
      uint16_t* base = q + ((long)row * hq + head) * d;
      const float c = cos_tab[p * half + j];
      const float s = sin_tab[p * half + j];
      const float x1 = __uint_as_float((uint32_t)base[2 * j] << 16);
      const float x2 = __uint_as_float((uint32_t)base[2 * j + 1] << 16);
      base[2 * j] = __bfloat16_as_ushort(
          __float2bfloat16((x1 * c - x2 * s) * q_scale));
      base[2 * j + 1] = __bfloat16_as_ushort(
          __float2bfloat16((x1 * s + x2 * c) * q_scale));
    } else if (head < hq + hk) {  // k: rotate -> cache
      const int h = head - hq;
      const uint16_t* base = k + ((long)row * hk + h) * d;
      uint16_t* dst =
          ck + (((long)row * hk + h) * cache_len + p) * d;
      const float c = cos_tab[p * half + j];
      const float s = sin_tab[p * half + j];
      const float x1 = __uint_as_float((uint32_t)base[2 * j] << 16);
      const float x2 = __uint_as_float((uint32_t)base[2 * j + 1] << 16);
      dst[2 * j] =
          __bfloat16_as_ushort(__float2bfloat16(x1 * c - x2 * s));
      dst[2 * j + 1] =
          __bfloat16_as_ushort(__float2bfloat16(x1 * s + x2 * c));
    } else {  // v: straight copy -> cache
      const int h = head - hq - hk;
      const uint16_t* base = v + ((long)row * hk + h) * d;
      uint16_t* dst =
          cv + (((long)row * hk + h) * cache_len + p) * d;
      dst[2 * j] = base[2 * j];
      dst[2 * j + 1] = base[2 * j + 1];
    }
  }
}

// Scalar fallbacks for pointers not 16-byte aligned (region offsets are
// caller-controlled; hipMalloc bases are 256-B aligned so the vector
// path is the common case).
extern "C" __global__ void cast_fp32_bf16_scalar_kernel(
    const uint32_t* __restrict__ src, uint16_t* __restrict__ dst, long n) {
  long i = (long)(blockIdx.x * blockDim.x + threadIdx.x);
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] = (uint16_t)(src[i] >> 16);
}

extern "C" __global__ void cast_bf16_fp32_scalar_kernel(
    const uint16_t* __restrict__ src, uint32_t* __restrict__ dst, long n) {
  long i = (long)(blockIdx.x * blockDim.x + threadIdx.x);
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] = (uint32_t)src[i] << 16;
}

extern "C" __global__ void cast_fp32_fp8e4m3_scalar_kernel(
    const float* __restrict__ src, uint8_t* __restrict__ dst, long n) {
  long i = (long)(blockIdx.x * blockDim.x + threadIdx.x);
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] = __hip_fp8_e4m3(src[i]).__x;
}

extern "C" __global__ void cast_fp8e4m3_fp32_scalar_kernel(
    const uint8_t* __restrict__ src, float* __restrict__ dst, long n) {
  long i = (long)(blockIdx.x * blockDim.x + threadIdx.x);
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    __hip_fp8_e4m3 v;
    v.__x = src[i];
    dst[i] = float(v);
  }
}

// Strided -> contiguous gather (up to 4-D), element size 1/2/4/8 bytes.
// Lifts the reference's "DLPack tensor must be contiguous" restriction
// (cuda_shared_memory/__init__.py:345-352) with a device-side pack.
template <typename T>
__global__ void gather_pack_kernel(const char* __restrict__ src,
                                   T* __restrict__ dst, long n,
                                   long s0, long s1, long s2, long s3,
                                   long d0, long d1, long d2, long d3) {
  long i = (long)(blockIdx.x * blockDim.x + threadIdx.x);
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    long rem = i;
    long i3 = rem % d3; rem /= d3;
    long i2 = rem % d2; rem /= d2;
    long i1 = rem % d1; rem /= d1;
    long i0 = rem;
    const char* p = src + ((i0 * s0 + i1 * s1 + i2 * s2 + i3 * s3) * (long)sizeof(T));
    dst[i] = *reinterpret_cast<const T*>(p);
  }
}

// LDS-tiled batched transpose: the gather_pack fast path for the
// pattern "innermost output dim is strided in the source, the dim
// before it is contiguous" (a transpose view, the worst case for the
// naive per-element kernel: 729 GB/s vs ~5 TB/s for contiguous casts —
// profiles/kernels_rocprof_r01.txt:18). 64x64 tile staged through LDS:
// loads coalesce along the source's contiguous dim, stores along the
// destination's, both full 64-lane wavefronts. +1 column pad keeps the
// LDS column reads conflict-free.
//
//   dst[b0][b1][r][c] = src[b0*s0 + b1*s1 + r + c*col_stride]
//   (r = contiguous source dim of extent rows; c of extent cols;
//    blockIdx.z enumerates the flattened (b0, b1) outer batch)
#define CA_TILE 64
template <typename T>
__global__ void transpose_tiled_kernel(const T* __restrict__ src,
                                       T* __restrict__ dst, long rows,
                                       long cols, long col_stride, long s0,
                                       long s1, long d1) {
  __shared__ T tile[CA_TILE][CA_TILE + 1];
  const long batch = blockIdx.z;
  const T* sb = src + (batch / d1) * s0 + (batch % d1) * s1;
  T* db = dst + batch * rows * cols;
  const long r0 = (long)blockIdx.x * CA_TILE;  // along rows (src-contig)
  const long c0 = (long)blockIdx.y * CA_TILE;  // along cols (dst-contig)
  const int tx = threadIdx.x % CA_TILE;        // fast lane index
  const int ty0 = threadIdx.x / CA_TILE;       // 4 rows/pass with 256 thr
  // load: lanes sweep the contiguous source dim (r), rows of the tile
  // are different c values
  for (int ty = ty0; ty < CA_TILE; ty += blockDim.x / CA_TILE) {
    const long c = c0 + ty;
    const long r = r0 + tx;
    if (r < rows && c < cols) {
      tile[ty][tx] = sb[r + c * col_stride];
    }
  }
  __syncthreads();
  // store: lanes sweep the contiguous destination dim (c)
  for (int ty = ty0; ty < CA_TILE; ty += blockDim.x / CA_TILE) {
    const long r = r0 + ty;
    const long c = c0 + tx;
    if (r < rows && c < cols) {
      db[r * cols + c] = tile[tx][ty];
    }
  }
}

// 2-byte specialization of the tiled transpose: each lane moves a PAIR
// of elements (u32) in both phases, restoring full bus width (the
// generic u16 path measured 2.3 TB/s vs 4.2 for fp32). Layout per
// 64x64 tile: load phase writes u32 pairs along the source-contiguous
// dim; store phase reads two adjacent tile rows and packs a u32 along
// the destination-contiguous dim. Only used for fully-covered tiles;
// ragged edges fall back to the scalar path in the launcher.
extern "C" __global__ void transpose_tiled_pair16_kernel(
    const uint16_t* __restrict__ src, uint16_t* __restrict__ dst, long rows,
    long cols, long col_stride, long s0, long s1, long d1) {
  __shared__ uint16_t tile[CA_TILE][CA_TILE + 2];
  const long batch = blockIdx.z;
  const uint16_t* sb = src + (batch / d1) * s0 + (batch % d1) * s1;
  uint16_t* db = dst + batch * rows * cols;
  const long r0 = (long)blockIdx.x * CA_TILE;
  const long c0 = (long)blockIdx.y * CA_TILE;
  const int tp = threadIdx.x % 32;   // pair index along the fast dim
  const int ty0 = threadIdx.x / 32;  // 8 slow-dim rows per pass
  // load: pairs along r (source-contiguous)
  for (int ty = ty0; ty < CA_TILE; ty += 8) {
    const long c = c0 + ty;
    const long r = r0 + 2 * tp;
    if (c < cols && r + 1 < rows) {
      const uint32_t v = *reinterpret_cast<const uint32_t*>(
          sb + r + c * col_stride);
      *reinterpret_cast<uint32_t*>(&tile[ty][2 * tp]) = v;
    } else if (c < cols && r < rows) {
      tile[ty][2 * tp] = sb[r + c * col_stride];
    }
  }
  __syncthreads();
  // store: pairs along c (destination-contiguous)
  for (int ty = ty0; ty < CA_TILE; ty += 8) {
    const long r = r0 + ty;
    const long c = c0 + 2 * tp;
    if (r < rows && c + 1 < cols) {
      uint32_t v = (uint32_t)tile[2 * tp][ty] |
                   ((uint32_t)tile[2 * tp + 1][ty] << 16);
      *reinterpret_cast<uint32_t*>(db + r * cols + c) = v;
    } else if (r < rows && c < cols) {
      db[r * cols + c] = tile[2 * tp][ty];
    }
  }
}

// Image preprocess: u8 HWC (ih,iw,3) -> bilinear resize (oh,ow) ->
// normalize -> planar CHW fp32 (or bf16). One thread per output pixel
// computes all 3 channels (reads coalesce along ow; the 4 source pixels
// hit L1/L2 for neighbors). Modes: 0 = raw /1, 1 = INCEPTION
// (x/127.5 - 1), 2 = VGG (x - mean_c), matching image_client.cc:86-190.
extern "C" __global__ void image_preprocess_kernel(
    const uint8_t* __restrict__ src, float* __restrict__ dst,
    int ih, int iw, int oh, int ow, int mode, int out_bf16,
    float m0, float m1, float m2, float s0, float s1, float s2) {
  long n = (long)oh * ow;
  long i = (long)(blockIdx.x * blockDim.x + threadIdx.x);
  long stride = (long)gridDim.x * blockDim.x;
  float scale_h = (float)ih / oh;
  float scale_w = (float)iw / ow;
  for (; i < n; i += stride) {
    int oy = i / ow;
    int ox = i % ow;
    float fy = (oy + 0.5f) * scale_h - 0.5f;
    float fx = (ox + 0.5f) * scale_w - 0.5f;
    int y0 = max(0, (int)floorf(fy));
    int x0 = max(0, (int)floorf(fx));
    int y1 = min(ih - 1, y0 + 1);
    int x1 = min(iw - 1, x0 + 1);
    y0 = min(y0, ih - 1);
    x0 = min(x0, iw - 1);
    float wy = fy - floorf(fy);
    float wx = fx - floorf(fx);
    if (fy < 0) wy = 0.f;
    if (fx < 0) wx = 0.f;
    const uint8_t* p00 = src + ((long)y0 * iw + x0) * 3;
    const uint8_t* p01 = src + ((long)y0 * iw + x1) * 3;
    const uint8_t* p10 = src + ((long)y1 * iw + x0) * 3;
    const uint8_t* p11 = src + ((long)y1 * iw + x1) * 3;
#pragma unroll
    for (int c = 0; c < 3; ++c) {
      float v = (1 - wy) * ((1 - wx) * p00[c] + wx * p01[c]) +
                wy * ((1 - wx) * p10[c] + wx * p11[c]);
      float mean = c == 0 ? m0 : (c == 1 ? m1 : m2);
      float sc = c == 0 ? s0 : (c == 1 ? s1 : s2);
      if (mode == 1) {
        v = v / 127.5f - 1.0f;
      } else if (mode == 2) {
        v = v - mean;
      } else {
        v = (v - mean) * sc;
      }
      long out_idx = (long)c * n + i;  // CHW
      if (out_bf16) {
        reinterpret_cast<uint16_t*>(dst)[out_idx] =
            (uint16_t)(__float_as_uint(v) >> 16);
      } else {
        dst[out_idx] = v;
      }
    }
  }
}


// Skinny decode GEMM: y[8, N] = x[8, K] @ W[N, K]^T, bf16 in/out,
// fp32 accumulate. The decode batch is tiny (8 rows), so the GEMM is
// pure weight streaming — hipBLASLt's skinny-tile kernels measured
// 2.1-4.2 TB/s on these shapes (profiles/decode_rocprof_r02.txt);
// this kernel is weights-stationary: x lives in LDS (staged in K
// chunks), W streams once with 16-B coalesced lane loads, each lane
// keeps 8 fp32 accumulators (one per batch row), cross-lane reduce at
// the end. One wavefront per W row, 4 rows per workgroup, grid-stride
// over N.
#define DG_M 8          // decode batch rows (compile-time; pad to 8)
#define DG_KC 4096      // K chunk staged in LDS: 8*4096*2 B = 64 KB
extern "C" __global__ void __launch_bounds__(256)
decode_gemm_bf16_kernel(const uint16_t* __restrict__ x,
                        const uint16_t* __restrict__ w,
                        uint16_t* __restrict__ y, int N, int K) {
  __shared__ uint16_t xs[DG_M * DG_KC];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;  // 4 waves = 4 W rows
  float acc[DG_M];
#pragma unroll
  for (int m = 0; m < DG_M; ++m) acc[m] = 0.f;

  const long row0 = (long)blockIdx.x * 4;
  // grid-stride over row groups so any N works with a capped grid
  for (long nb = row0; nb < N; nb += (long)gridDim.x * 4) {
    const long n = nb + wave;
#pragma unroll
    for (int m = 0; m < DG_M; ++m) acc[m] = 0.f;
    for (int kc = 0; kc < K; kc += DG_KC) {
      const int kn = min(DG_KC, K - kc);
      // cooperative stage of x[:, kc:kc+kn] (row-major [M, K])
      __syncthreads();
      for (int idx = threadIdx.x * 8; idx < DG_M * kn;
           idx += blockDim.x * 8) {
        const int m = idx / kn;
        const int k = idx - m * kn;
        if (k + 8 <= kn) {
          *reinterpret_cast<uint4*>(xs + m * DG_KC + k) =
              *reinterpret_cast<const uint4*>(x + (long)m * K + kc + k);
        } else {
          for (int t = k; t < kn; ++t)
            xs[m * DG_KC + t] = x[(long)m * K + kc + t];
        }
      }
      __syncthreads();
      if (n < N) {
        const uint16_t* wr = w + n * (long)K + kc;
        // lane streams 8 W elems per iteration, 64 lanes cover 512
        for (int k0 = lane * 8; k0 + 8 <= kn; k0 += 64 * 8) {
          uint4 wv = *reinterpret_cast<const uint4*>(wr + k0);
          const uint16_t* we = reinterpret_cast<const uint16_t*>(&wv);
          float wf[8];
#pragma unroll
          for (int j = 0; j < 8; ++j)
            wf[j] = __uint_as_float((uint32_t)we[j] << 16);
#pragma unroll
          for (int m = 0; m < DG_M; ++m) {
            const uint16_t* xr = xs + m * DG_KC + k0;
            uint4 xv = *reinterpret_cast<const uint4*>(xr);
            const uint16_t* xe = reinterpret_cast<const uint16_t*>(&xv);
            float s = 0.f;
#pragma unroll
            for (int j = 0; j < 8; ++j)
              s = fmaf(__uint_as_float((uint32_t)xe[j] << 16), wf[j], s);
            acc[m] += s;
          }
        }
        // (no ragged-K handling: the launcher rejects K % 512 != 0)
      }
    }
    if (n < N) {
      // cross-lane reduce each row's 8 accumulators
#pragma unroll
      for (int m = 0; m < DG_M; ++m) {
        float v = acc[m];
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off);
        if (lane == 0) {
          uint32_t u = __float_as_uint(v);
          u += 0x7FFF + ((u >> 16) & 1);  // RNE to bf16
          y[(long)m * N + n] = (uint16_t)(u >> 16);
        }
      }
    }
  }
}

extern "C" hipError_t ca_decode_gemm_bf16(const void* x, const void* w,
                                          void* y, int N, int K,
                                          hipStream_t stream) {
  // K must be a multiple of 512 (64 lanes x 8) — true for every
  // decode GEMM in the model family (4096, 14336, 1024); reject others
  // so callers fall back to torch.
  if (K % 512 != 0) return hipErrorInvalidValue;
  if (((uintptr_t)x | (uintptr_t)w | (uintptr_t)y) & 15)
    return hipErrorInvalidValue;
  long groups = ((long)N + 3) / 4;
  int grid = (int)(groups > 2048 ? 2048 : groups);
  hipLaunchKernelGGL(decode_gemm_bf16_kernel, dim3(grid), dim3(256), 0,
                     stream, (const uint16_t*)x, (const uint16_t*)w,
                     (uint16_t*)y, N, K);
  return hipGetLastError();
}

// ---------------------------------------------------------------------------
// C-ABI launchers (alignment dispatch included). Return hipError_t.
// ---------------------------------------------------------------------------

extern "C" hipError_t ca_cast_fp32_bf16(const void* src, void* dst, long n,
                                        hipStream_t stream) {
  auto s = reinterpret_cast<uintptr_t>(src);
  auto d = reinterpret_cast<uintptr_t>(dst);
  if ((s | d) & 15) {
    hipLaunchKernelGGL(cast_fp32_bf16_scalar_kernel, dim3(ca_grid_for(n)),
                       dim3(256), 0, stream, (const uint32_t*)src,
                       (uint16_t*)dst, n);
  } else {
    hipLaunchKernelGGL(cast_fp32_bf16_v2_kernel,
                       dim3(ca_grid_for((n + 15) / 16)), dim3(256), 0, stream,
                       (const uint32_t*)src, (uint16_t*)dst, n);
  }
  return hipGetLastError();
}

extern "C" hipError_t ca_cast_bf16_fp32(const void* src, void* dst, long n,
                                        hipStream_t stream) {
  auto s = reinterpret_cast<uintptr_t>(src);
  auto d = reinterpret_cast<uintptr_t>(dst);
  if ((s | d) & 15) {
    hipLaunchKernelGGL(cast_bf16_fp32_scalar_kernel, dim3(ca_grid_for(n)),
                       dim3(256), 0, stream, (const uint16_t*)src,
                       (uint32_t*)dst, n);
  } else {
    hipLaunchKernelGGL(cast_bf16_fp32_kernel, dim3(ca_grid_for((n + 7) / 8)),
                       dim3(256), 0, stream, (const uint16_t*)src,
                       (uint32_t*)dst, n);
  }
  return hipGetLastError();
}

extern "C" hipError_t ca_cast_fp32_fp8e4m3(const void* src, void* dst, long n,
                                           hipStream_t stream) {
  auto s = reinterpret_cast<uintptr_t>(src);
  auto d = reinterpret_cast<uintptr_t>(dst);
  if ((s | d) & 15) {
    hipLaunchKernelGGL(cast_fp32_fp8e4m3_scalar_kernel, dim3(ca_grid_for(n)),
                       dim3(256), 0, stream, (const float*)src,
                       (uint8_t*)dst, n);
  } else {
    hipLaunchKernelGGL(cast_fp32_fp8e4m3_kernel,
                       dim3(ca_grid_for((n + 7) / 8)), dim3(256), 0, stream,
                       (const float*)src, (uint8_t*)dst, n);
  }
  return hipGetLastError();
}

extern "C" hipError_t ca_cast_fp8e4m3_fp32(const void* src, void* dst, long n,
                                           hipStream_t stream) {
  auto s = reinterpret_cast<uintptr_t>(src);
  auto d = reinterpret_cast<uintptr_t>(dst);
  if ((s | d) & 15) {
    hipLaunchKernelGGL(cast_fp8e4m3_fp32_scalar_kernel, dim3(ca_grid_for(n)),
                       dim3(256), 0, stream, (const uint8_t*)src,
                       (float*)dst, n);
  } else {
    hipLaunchKernelGGL(cast_fp8e4m3_fp32_kernel,
                       dim3(ca_grid_for((n + 7) / 8)), dim3(256), 0, stream,
                       (const uint8_t*)src, (float*)dst, n);
  }
  return hipGetLastError();
}

namespace {

template <typename T>
hipError_t ca_transpose_tiled_launch(const void* src, void* dst,
                                     const long* shape, const long* strides,
                                     hipStream_t stream) {
  const long rows = shape[2], cols = shape[3];
  dim3 grid((uint32_t)((rows + CA_TILE - 1) / CA_TILE),
            (uint32_t)((cols + CA_TILE - 1) / CA_TILE),
            (uint32_t)(shape[0] * shape[1]));
  hipLaunchKernelGGL((transpose_tiled_kernel<T>), grid, dim3(256), 0, stream,
                     (const T*)src, (T*)dst, rows, cols, strides[3],
                     strides[0], strides[1], shape[1]);
  return hipGetLastError();
}

}  // namespace

extern "C" hipError_t ca_gather_pack(const void* src, void* dst,
                                     int elem_size, const long* shape,
                                     const long* strides,
                                     hipStream_t stream) {
  // shape/strides are 4-element host arrays (leading dims padded)
  long n = shape[0] * shape[1] * shape[2] * shape[3];
  // transpose-like fast path: dim 2 contiguous in the source, innermost
  // output dim strided -> LDS-tiled transpose (coalesced both ways;
  // the naive per-element path was 7x off peak on this pattern,
  // profiles/kernels_rocprof_r01.txt:18)
  if (strides[2] == 1 && strides[3] != 1 && shape[2] >= 16 &&
      shape[3] >= 16 && shape[0] * shape[1] <= 65535 &&
      (shape[3] + CA_TILE - 1) / CA_TILE <= 65535) {
    switch (elem_size) {
      case 1:
        return ca_transpose_tiled_launch<uint8_t>(src, dst, shape, strides,
                                                  stream);
      case 2: {
        // pair-per-lane variant restores full bus width for 2-byte
        // elements (u16 path: 2.3 TB/s; fp32: 4.2)
        const long rows = shape[2], cols = shape[3];
        dim3 grid((uint32_t)((rows + CA_TILE - 1) / CA_TILE),
                  (uint32_t)((cols + CA_TILE - 1) / CA_TILE),
                  (uint32_t)(shape[0] * shape[1]));
        hipLaunchKernelGGL(transpose_tiled_pair16_kernel, grid, dim3(256),
                           0, stream, (const uint16_t*)src, (uint16_t*)dst,
                           rows, cols, strides[3], strides[0], strides[1],
                           shape[1]);
        return hipGetLastError();
      }
      case 4:
        return ca_transpose_tiled_launch<uint32_t>(src, dst, shape, strides,
                                                   stream);
      case 8:
        return ca_transpose_tiled_launch<uint64_t>(src, dst, shape, strides,
                                                   stream);
      default:
        break;  // fall through to the scalar path
    }
  }
  int grid = ca_grid_for(n);
  const char* sp = (const char*)src;
  switch (elem_size) {
    case 1:
      hipLaunchKernelGGL((gather_pack_kernel<uint8_t>), dim3(grid), dim3(256),
                         0, stream, sp, (uint8_t*)dst, n, strides[0],
                         strides[1], strides[2], strides[3], shape[0],
                         shape[1], shape[2], shape[3]);
      break;
    case 2:
      hipLaunchKernelGGL((gather_pack_kernel<uint16_t>), dim3(grid), dim3(256),
                         0, stream, sp, (uint16_t*)dst, n, strides[0],
                         strides[1], strides[2], strides[3], shape[0],
                         shape[1], shape[2], shape[3]);
      break;
    case 4:
      hipLaunchKernelGGL((gather_pack_kernel<uint32_t>), dim3(grid), dim3(256),
                         0, stream, sp, (uint32_t*)dst, n, strides[0],
                         strides[1], strides[2], strides[3], shape[0],
                         shape[1], shape[2], shape[3]);
      break;
    case 8:
      hipLaunchKernelGGL((gather_pack_kernel<uint64_t>), dim3(grid), dim3(256),
                         0, stream, sp, (uint64_t*)dst, n, strides[0],
                         strides[1], strides[2], strides[3], shape[0],
                         shape[1], shape[2], shape[3]);
      break;
    default:
      return hipErrorInvalidValue;
  }
  return hipGetLastError();
}

extern "C" hipError_t ca_image_preprocess(const void* src, void* dst, int ih,
                                          int iw, int oh, int ow, int mode,
                                          int out_bf16, const float* mean,
                                          const float* stdev,
                                          hipStream_t stream) {
  hipLaunchKernelGGL(image_preprocess_kernel,
                     dim3(ca_grid_for((long)oh * ow)), dim3(256), 0, stream,
                     (const uint8_t*)src, (float*)dst, ih, iw, oh, ow, mode,
                     out_bf16, mean[0], mean[1], mean[2], stdev[0], stdev[1],
                     stdev[2]);
  return hipGetLastError();
}

// Batched variant: N same-sized images in one launch (u8 HWC stacked
// contiguously -> NCHW). The single-image kernel is launch-bound at
// high request rates (~27 us/image vs ~2-3 us of kernel time for
// 224x224 — profiles/kernels_rocprof_r01.txt:19); the batch amortizes
// one launch over the whole request. blockIdx.y = image index (grid.y
// caps at 65535 — far above any sane batch).
extern "C" __global__ void image_preprocess_batched_kernel(
    const uint8_t* __restrict__ src, float* __restrict__ dst, int ih, int iw,
    int oh, int ow, int mode, int out_bf16, float m0, float m1, float m2,
    float s0, float s1, float s2) {
  const long in_img = (long)ih * iw * 3;
  const long n = (long)oh * ow;
  const uint8_t* simg = src + (long)blockIdx.y * in_img;
  float* dimg = dst + (long)blockIdx.y * 3 * n;
  uint16_t* dimg16 =
      reinterpret_cast<uint16_t*>(dst) + (long)blockIdx.y * 3 * n;
  long i = (long)(blockIdx.x * blockDim.x + threadIdx.x);
  const long stride = (long)gridDim.x * blockDim.x;
  const float scale_h = (float)ih / oh;
  const float scale_w = (float)iw / ow;
  for (; i < n; i += stride) {
    int oy = i / ow;
    int ox = i % ow;
    float fy = (oy + 0.5f) * scale_h - 0.5f;
    float fx = (ox + 0.5f) * scale_w - 0.5f;
    int y0 = max(0, (int)floorf(fy));
    int x0 = max(0, (int)floorf(fx));
    int y1 = min(ih - 1, y0 + 1);
    int x1 = min(iw - 1, x0 + 1);
    y0 = min(y0, ih - 1);
    x0 = min(x0, iw - 1);
    float wy = fy - floorf(fy);
    float wx = fx - floorf(fx);
    if (fy < 0) wy = 0.f;
    if (fx < 0) wx = 0.f;
    const uint8_t* p00 = simg + ((long)y0 * iw + x0) * 3;
    const uint8_t* p01 = simg + ((long)y0 * iw + x1) * 3;
    const uint8_t* p10 = simg + ((long)y1 * iw + x0) * 3;
    const uint8_t* p11 = simg + ((long)y1 * iw + x1) * 3;
#pragma unroll
    for (int c = 0; c < 3; ++c) {
      float v = (1 - wy) * ((1 - wx) * p00[c] + wx * p01[c]) +
                wy * ((1 - wx) * p10[c] + wx * p11[c]);
      float mean = c == 0 ? m0 : (c == 1 ? m1 : m2);
      float sc = c == 0 ? s0 : (c == 1 ? s1 : s2);
      if (mode == 1) {
        v = v / 127.5f - 1.0f;
      } else if (mode == 2) {
        v = v - mean;
      } else {
        v = (v - mean) * sc;
      }
      long out_idx = (long)c * n + i;  // CHW within the image
      if (out_bf16) {
        dimg16[out_idx] = (uint16_t)(__float_as_uint(v) >> 16);
      } else {
        dimg[out_idx] = v;
      }
    }
  }
}

extern "C" hipError_t ca_image_preprocess_batched(
    const void* src, void* dst, int n_images, int ih, int iw, int oh, int ow,
    int mode, int out_bf16, const float* mean, const float* stdev,
    hipStream_t stream) {
  if (n_images < 1 || n_images > 65535) return hipErrorInvalidValue;
  dim3 grid((uint32_t)ca_grid_for((long)oh * ow), (uint32_t)n_images);
  hipLaunchKernelGGL(image_preprocess_batched_kernel, grid, dim3(256), 0,
                     stream, (const uint8_t*)src, (float*)dst, ih, iw, oh,
                     ow, mode, out_bf16, mean[0], mean[1], mean[2], stdev[0],
                     stdev[1], stdev[2]);
  return hipGetLastError();
}

// Per-channel bias add over an NCHW bf16 tensor, in-place capable —
// absorbs the separate elementwise bias pass torch-on-ROCm emits for
// conv biases created by BatchNorm folding (55 ms / 5247 calls in the
// ResNet50 serving profile). One block per (n,c) plane: the fp32 bias
// loads once, elements stream 8-wide with a scalar tail so any plane
// size (incl. 7x7=49) is handled. Optional fused ReLU.
extern "C" __global__ void bias_act_bf16_kernel(
    const uint16_t* __restrict__ x, const float* __restrict__ bias,
    uint16_t* __restrict__ out, long plane, int channels, int do_relu) {
  const long pidx = blockIdx.x;            // n*C + c
  const int c = (int)(pidx % channels);
  const float b = bias[c];
  const uint16_t* xp = x + pidx * plane;
  uint16_t* op = out + pidx * plane;
  const long vec_n = plane / 8;

  auto apply = [&](uint16_t v) -> uint16_t {
    float f = __uint_as_float(((uint32_t)v) << 16) + b;
    if (do_relu && f < 0.f) f = 0.f;
    // round-to-nearest-even bf16 (matches torch float->bf16 casts)
    uint32_t u = __float_as_uint(f);
    u += 0x7FFF + ((u >> 16) & 1);
    return (uint16_t)(u >> 16);
  };

  const uint4* xv = (const uint4*)xp;
  uint4* ov = (uint4*)op;
  for (long i = threadIdx.x; i < vec_n; i += blockDim.x) {
    uint4 v = xv[i];
    uint32_t* w = (uint32_t*)&v;
    for (int j = 0; j < 4; ++j) {
      uint16_t lo = apply((uint16_t)(w[j] & 0xFFFF));
      uint16_t hi = apply((uint16_t)(w[j] >> 16));
      w[j] = (uint32_t)lo | ((uint32_t)hi << 16);
    }
    ov[i] = v;
  }
  for (long i = vec_n * 8 + threadIdx.x; i < plane; i += blockDim.x) {
    op[i] = apply(xp[i]);
  }
}

extern "C" hipError_t ca_bias_act_bf16(const void* x, const void* bias,
                                       void* out, long n_planes, long plane,
                                       int channels, int do_relu,
                                       hipStream_t stream) {
  if (((uintptr_t)x & 15) || ((uintptr_t)out & 15)) {
    return hipErrorInvalidValue;  // 16B alignment for the uint4 path
  }
  hipLaunchKernelGGL(bias_act_bf16_kernel, dim3((uint32_t)n_planes),
                     dim3(256), 0, stream, (const uint16_t*)x,
                     (const float*)bias, (uint16_t*)out, plane, channels,
                     do_relu);
  return hipGetLastError();
}

// Per-channel bias + residual add + ReLU over NCHW bf16 in ONE pass —
// the bottleneck-exit fusion (out = relu(x + bias[c] + residual)).
// After BN folding, torch-on-ROCm runs this as three elementwise
// kernels (bias add, residual add, relu), i.e. 3 extra full memory
// passes over the activation per block exit; this kernel is one
// read-x + read-residual + write pass. fp32 math, RNE back to bf16.
// residual may be null (degenerates to bias[+relu]).
extern "C" __global__ void bias_res_act_bf16_kernel(
    const uint16_t* __restrict__ x, const uint16_t* __restrict__ res,
    const float* __restrict__ bias, uint16_t* __restrict__ out, long plane,
    int channels, int do_relu) {
  const long pidx = blockIdx.x;  // n*C + c
  const int c = (int)(pidx % channels);
  const float b = bias[c];
  const uint16_t* xp = x + pidx * plane;
  const uint16_t* rp = res ? res + pidx * plane : nullptr;
  uint16_t* op = out + pidx * plane;
  const long vec_n = plane / 8;

  auto fin = [&](float f) -> uint16_t {
    if (do_relu && f < 0.f) f = 0.f;
    uint32_t u = __float_as_uint(f);
    u += 0x7FFF + ((u >> 16) & 1);  // RNE to bf16
    return (uint16_t)(u >> 16);
  };

  const uint4* xv = (const uint4*)xp;
  const uint4* rv = (const uint4*)rp;
  uint4* ov = (uint4*)op;
  for (long i = threadIdx.x; i < vec_n; i += blockDim.x) {
    uint4 v = xv[i];
    uint4 r = rp ? rv[i] : uint4{0, 0, 0, 0};
    uint32_t* w = (uint32_t*)&v;
    const uint32_t* rw = (const uint32_t*)&r;
    for (int j = 0; j < 4; ++j) {
      // addition order (x + residual) + bias matches the torch
      // reference sequence exactly (fp32 add is non-associative;
      // a different order flips RNE boundary cases)
      float lo = __uint_as_float((w[j] & 0xFFFFu) << 16);
      float hi = __uint_as_float(w[j] & 0xFFFF0000u);
      if (rp) {
        lo += __uint_as_float((rw[j] & 0xFFFFu) << 16);
        hi += __uint_as_float(rw[j] & 0xFFFF0000u);
      }
      w[j] = (uint32_t)fin(lo + b) | ((uint32_t)fin(hi + b) << 16);
    }
    ov[i] = v;
  }
  for (long i = vec_n * 8 + threadIdx.x; i < plane; i += blockDim.x) {
    float f = __uint_as_float(((uint32_t)xp[i]) << 16);
    if (rp) f += __uint_as_float(((uint32_t)rp[i]) << 16);
    op[i] = fin(f + b);
  }
}

// Channels-LAST (NHWC-contiguous) variant: channel is the fastest
// dim, so each lane's 8 consecutive elements are 8 consecutive
// channels — loads of x, residual AND bias all coalesce. This is what
// lets the fused epilogues ride MIOpen's native NHWC conv path
// (channels_last measured catastrophic in r01 precisely because the
// NCHW-plane kernels fell back to eager torch ops on NHWC tensors).
// Requires channels % 8 == 0 (ResNet: 64..2048, all satisfy).
extern "C" __global__ void bias_res_act_cl_bf16_kernel(
    const uint16_t* __restrict__ x, const uint16_t* __restrict__ res,
    const float* __restrict__ bias, uint16_t* __restrict__ out, long n,
    int channels, int do_relu) {
  auto fin = [&](float f) -> uint16_t {
    if (do_relu && f < 0.f) f = 0.f;
    uint32_t u = __float_as_uint(f);
    u += 0x7FFF + ((u >> 16) & 1);  // RNE to bf16
    return (uint16_t)(u >> 16);
  };
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i + 8 <= n; i += stride) {
    const int c0 = (int)(i % channels);
    uint4 v = *reinterpret_cast<const uint4*>(x + i);
    uint4 r = res ? *reinterpret_cast<const uint4*>(res + i)
                  : uint4{0, 0, 0, 0};
    const float4 b0 = *reinterpret_cast<const float4*>(bias + c0);
    const float4 b1 = *reinterpret_cast<const float4*>(bias + c0 + 4);
    const float bv[8] = {b0.x, b0.y, b0.z, b0.w, b1.x, b1.y, b1.z, b1.w};
    uint32_t* w = (uint32_t*)&v;
    const uint32_t* rw = (const uint32_t*)&r;
    uint16_t o[8];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float lo = __uint_as_float((w[j] & 0xFFFFu) << 16);
      float hi = __uint_as_float(w[j] & 0xFFFF0000u);
      if (res) {
        lo += __uint_as_float((rw[j] & 0xFFFFu) << 16);
        hi += __uint_as_float(rw[j] & 0xFFFF0000u);
      }
      o[2 * j] = fin(lo + bv[2 * j]);
      o[2 * j + 1] = fin(hi + bv[2 * j + 1]);
    }
    *reinterpret_cast<uint4*>(out + i) = *reinterpret_cast<uint4*>(o);
  }
  // tail (n % 8 == 0 whenever channels % 8 == 0, but keep it safe)
  long tail = (n / 8) * 8;
  long ti = tail + (blockIdx.x * blockDim.x + threadIdx.x);
  if (ti < n && (blockIdx.x * blockDim.x + threadIdx.x) < 8) {
    float f = __uint_as_float(((uint32_t)x[ti]) << 16);
    if (res) f += __uint_as_float(((uint32_t)res[ti]) << 16);
    out[ti] = fin(f + bias[ti % channels]);
  }
}

extern "C" hipError_t ca_bias_res_act_cl_bf16(const void* x, const void* res,
                                              const void* bias, void* out,
                                              long n, int channels,
                                              int do_relu,
                                              hipStream_t stream) {
  if (channels % 8 != 0) return hipErrorInvalidValue;
  if (((uintptr_t)x & 15) || ((uintptr_t)out & 15) ||
      ((uintptr_t)res & 15)) {
    return hipErrorInvalidValue;
  }
  hipLaunchKernelGGL(bias_res_act_cl_bf16_kernel,
                     dim3(ca_grid_for((n + 7) / 8)), dim3(256), 0, stream,
                     (const uint16_t*)x, (const uint16_t*)res,
                     (const float*)bias, (uint16_t*)out, n, channels,
                     do_relu);
  return hipGetLastError();
}

extern "C" hipError_t ca_bias_res_act_bf16(const void* x, const void* res,
                                           const void* bias, void* out,
                                           long n_planes, long plane,
                                           int channels, int do_relu,
                                           hipStream_t stream) {
  if (((uintptr_t)x & 15) || ((uintptr_t)out & 15) ||
      ((uintptr_t)res & 15)) {
    return hipErrorInvalidValue;  // 16B alignment for the uint4 path
  }
  hipLaunchKernelGGL(bias_res_act_bf16_kernel, dim3((uint32_t)n_planes),
                     dim3(256), 0, stream, (const uint16_t*)x,
                     (const uint16_t*)res, (const float*)bias,
                     (uint16_t*)out, plane, channels, do_relu);
  return hipGetLastError();
}

extern "C" hipError_t ca_rmsnorm_bf16(const void* x, const void* w, void* out,
                                      long rows, int dim, float eps,
                                      hipStream_t stream) {
  if (dim % 8 != 0) return hipErrorInvalidValue;
  hipLaunchKernelGGL(rmsnorm_bf16_kernel, dim3((uint32_t)rows), dim3(256), 0,
                     stream, (const uint16_t*)x, (const uint16_t*)w,
                     (uint16_t*)out, dim, eps);
  return hipGetLastError();
}

extern "C" hipError_t ca_rope_decode_bf16(void* q, void* k,
                                          const void* cos_tab,
                                          const void* sin_tab,
                                          const void* pos, int b, int hq,
                                          int hk, int d, hipStream_t stream) {
  long total = (long)b * (hq + hk) * (d / 2);
  hipLaunchKernelGGL(rope_decode_bf16_kernel, dim3(ca_grid_for(total)),
                     dim3(256), 0, stream, (uint16_t*)q, (uint16_t*)k,
                     (const float*)cos_tab, (const float*)sin_tab,
                     (const long*)pos, b, hq, hk, d);
  return hipGetLastError();
}

extern "C" hipError_t ca_rope_scatter_decode_bf16(
    void* q, const void* k, const void* v, void* ck, void* cv,
    const void* cos_tab, const void* sin_tab, const void* pos, int b,
    int hq, int hk, int d, long cache_len, float q_scale,
    hipStream_t stream) {
  long total = (long)b * (hq + 2 * hk) * (d / 2);
  hipLaunchKernelGGL(rope_scatter_decode_bf16_kernel,
                     dim3(ca_grid_for(total)), dim3(256), 0, stream,
                     (uint16_t*)q, (const uint16_t*)k, (const uint16_t*)v,
                     (uint16_t*)ck, (uint16_t*)cv, (const float*)cos_tab,
                     (const float*)sin_tab, (const long*)pos, b, hq, hk, d,
                     cache_len, q_scale);
  return hipGetLastError();
}
