// client_amd._hip_c — MI355X-native HIP runtime + CDNA4 kernels.
//
// The GPU surface of the client stack: HIP-IPC shared
// memory (hipMalloc + hipIpcGetMemHandle / hipIpcOpenMemHandle), async
// copies on cached per-device streams, and hand-written gfx950 kernels
// for the work the reference does on the CPU (per-element BF16 loops,
// OpenCV preprocess):
//   - cast_fp32_bf16 / cast_bf16_fp32   (wire-exact truncate / zero-pad)
//   - cast_fp32_fp8e4m3 / cast_fp8e4m3_fp32 (OCP e4m3fn, CDNA4 native)
//   - gather_pack (strided -> contiguous, lifts the reference's DLPack
//     contiguity restriction, cuda_shared_memory/__init__.py:328-388)
//   - image_preprocess (u8 HWC -> resize bilinear -> normalize -> CHW,
//     replaces image_client.cc:86-190's OpenCV path)
//
// All kernels are memory-bound streaming kernels: 16 B/lane vectorized
// access, 256-thread blocks (4 waves of 64), grid-stride with the grid
// capped at 2048 workgroups (cdna_hip_programming.md §6 G11/G13).
// Pure HIP — no CUDA headers, no torch dependency; torch interop is via
// DLPack in Python.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstring>
#include <mutex>
#include <stdexcept>
#include <string>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

#define HIP_CHECK(expr)                                                    \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess) {                                                \
      throw std::runtime_error(std::string(#expr) + " failed: " +          \
                               hipGetErrorString(_e));                     \
    }                                                                      \
  } while (0)

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

// ---------------------------------------------------------------------------
// Launch helpers
// ---------------------------------------------------------------------------

static inline int grid_for(long work_items) {
  // memory-bound streaming grid: cap at 2048 workgroups, grid-stride the
  // rest (G11). 256 CUs x 8 blocks/CU.
  long blocks = (work_items + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

// Per-device cached streams: [0] = copy/pack stream, [1] = side stream for
// collective overlap (SURVEY.md §2.9 row 5 mandates the second stream).
static std::mutex g_stream_mu;
static std::unordered_map<int, std::vector<hipStream_t>> g_streams;

static hipStream_t get_stream(int device, int idx = 0) {
  std::lock_guard<std::mutex> lock(g_stream_mu);
  auto& vec = g_streams[device];
  while ((int)vec.size() <= idx) {
    int prev;
    HIP_CHECK(hipGetDevice(&prev));
    HIP_CHECK(hipSetDevice(device));
    hipStream_t s;
    HIP_CHECK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
    vec.push_back(s);
    HIP_CHECK(hipSetDevice(prev));
  }
  return vec[idx];
}

// ---------------------------------------------------------------------------
// Python bindings
// ---------------------------------------------------------------------------

static int device_count() {
  int n = 0;
  hipError_t e = hipGetDeviceCount(&n);
  if (e != hipSuccess) return 0;
  return n;
}

static uintptr_t hip_malloc(int device, size_t byte_size) {
  int prev;
  HIP_CHECK(hipGetDevice(&prev));
  HIP_CHECK(hipSetDevice(device));
  void* ptr = nullptr;
  HIP_CHECK(hipMalloc(&ptr, byte_size));
  HIP_CHECK(hipSetDevice(prev));
  return reinterpret_cast<uintptr_t>(ptr);
}

static void hip_free(uintptr_t ptr) {
  HIP_CHECK(hipFree(reinterpret_cast<void*>(ptr)));
}

static py::bytes ipc_get_mem_handle(uintptr_t ptr) {
  hipIpcMemHandle_t handle;
  HIP_CHECK(hipIpcGetMemHandle(&handle, reinterpret_cast<void*>(ptr)));
  return py::bytes(reinterpret_cast<const char*>(&handle), sizeof(handle));
}

static uintptr_t ipc_open_mem_handle(py::bytes raw) {
  std::string s = raw;
  if (s.size() < sizeof(hipIpcMemHandle_t)) {
    s.resize(sizeof(hipIpcMemHandle_t), '\0');
  }
  hipIpcMemHandle_t handle;
  std::memcpy(&handle, s.data(), sizeof(handle));
  void* ptr = nullptr;
  HIP_CHECK(hipIpcOpenMemHandle(&ptr, handle, hipIpcMemLazyEnablePeerAccess));
  return reinterpret_cast<uintptr_t>(ptr);
}

static void ipc_close_mem_handle(uintptr_t ptr) {
  HIP_CHECK(hipIpcCloseMemHandle(reinterpret_cast<void*>(ptr)));
}

static void memcpy_h2d(uintptr_t dst, py::buffer src, size_t byte_size,
                       int device, bool sync) {
  py::buffer_info info = src.request();
  if ((size_t)(info.size * info.itemsize) < byte_size)
    throw std::runtime_error("source buffer smaller than byte_size");
  hipStream_t s = get_stream(device);
  {
    py::gil_scoped_release release;
    HIP_CHECK(hipMemcpyAsync(reinterpret_cast<void*>(dst), info.ptr, byte_size,
                             hipMemcpyHostToDevice, s));
    if (sync) HIP_CHECK(hipStreamSynchronize(s));
  }
}

static py::bytes memcpy_d2h(uintptr_t src, size_t byte_size, int device) {
  std::string out(byte_size, '\0');
  hipStream_t s = get_stream(device);
  {
    py::gil_scoped_release release;
    HIP_CHECK(hipMemcpyAsync(&out[0], reinterpret_cast<void*>(src), byte_size,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
  }
  return py::bytes(out);
}

static void memcpy_d2h_into(uintptr_t src, py::buffer dst, size_t byte_size,
                            int device) {
  py::buffer_info info = dst.request(true);
  if ((size_t)(info.size * info.itemsize) < byte_size)
    throw std::runtime_error("destination buffer smaller than byte_size");
  hipStream_t s = get_stream(device);
  {
    py::gil_scoped_release release;
    HIP_CHECK(hipMemcpyAsync(info.ptr, reinterpret_cast<void*>(src), byte_size,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void memcpy_d2d(uintptr_t dst, uintptr_t src, size_t byte_size,
                       int device, bool sync) {
  hipStream_t s = get_stream(device);
  {
    py::gil_scoped_release release;
    HIP_CHECK(hipMemcpyAsync(reinterpret_cast<void*>(dst),
                             reinterpret_cast<void*>(src), byte_size,
                             hipMemcpyDeviceToDevice, s));
    if (sync) HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void stream_sync(int device, int idx) {
  hipStream_t s = get_stream(device, idx);
  py::gil_scoped_release release;
  HIP_CHECK(hipStreamSynchronize(s));
}

static void device_sync() {
  py::gil_scoped_release release;
  HIP_CHECK(hipDeviceSynchronize());
}

// ---- kernel wrappers (operate on raw device pointers) ----

static void cast_fp32_bf16(uintptr_t src, uintptr_t dst, long n, int device,
                           bool sync, int stream_idx) {
  hipStream_t s = get_stream(device, stream_idx);
  if ((src | dst) & 15) {
    hipLaunchKernelGGL(cast_fp32_bf16_scalar_kernel, dim3(grid_for(n)),
                       dim3(256), 0, s, reinterpret_cast<const uint32_t*>(src),
                       reinterpret_cast<uint16_t*>(dst), n);
  } else {
    // 16-elem/lane variant: measured 5222 vs 4941 GB/s for 8-elem
    // (profiles/kernels_rocprof_r01.txt A/B)
    hipLaunchKernelGGL(cast_fp32_bf16_v2_kernel, dim3(grid_for((n + 15) / 16)),
                       dim3(256), 0, s, reinterpret_cast<const uint32_t*>(src),
                       reinterpret_cast<uint16_t*>(dst), n);
  }
  HIP_CHECK(hipGetLastError());
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void cast_fp32_bf16_v2(uintptr_t src, uintptr_t dst, long n,
                              int device, bool sync, int stream_idx) {
  hipStream_t s = get_stream(device, stream_idx);
  int grid = grid_for((n + 15) / 16);
  hipLaunchKernelGGL(cast_fp32_bf16_v2_kernel, dim3(grid), dim3(256), 0, s,
                     reinterpret_cast<const uint32_t*>(src),
                     reinterpret_cast<uint16_t*>(dst), n);
  HIP_CHECK(hipGetLastError());
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void cast_fp32_bf16_v3(uintptr_t src, uintptr_t dst, long n,
                              int device, bool sync, int stream_idx) {
  hipStream_t s = get_stream(device, stream_idx);
  hipLaunchKernelGGL(cast_fp32_bf16_v3_kernel, dim3(grid_for((n + 15) / 16)),
                     dim3(256), 0, s, reinterpret_cast<const uint32_t*>(src),
                     reinterpret_cast<uint16_t*>(dst), n);
  HIP_CHECK(hipGetLastError());
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void cast_bf16_fp32(uintptr_t src, uintptr_t dst, long n, int device,
                           bool sync, int stream_idx) {
  hipStream_t s = get_stream(device, stream_idx);
  if ((src | dst) & 15) {
    hipLaunchKernelGGL(cast_bf16_fp32_scalar_kernel, dim3(grid_for(n)),
                       dim3(256), 0, s, reinterpret_cast<const uint16_t*>(src),
                       reinterpret_cast<uint32_t*>(dst), n);
  } else {
    // 8-elem/lane measured FASTER than 16-elem for the unpack
    // direction (4394 vs 4003 GB/s — write-heavy mix; opposite of the
    // pack kernel where 16-elem won)
    hipLaunchKernelGGL(cast_bf16_fp32_kernel, dim3(grid_for((n + 7) / 8)),
                       dim3(256), 0, s, reinterpret_cast<const uint16_t*>(src),
                       reinterpret_cast<uint32_t*>(dst), n);
  }
  HIP_CHECK(hipGetLastError());
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void cast_fp32_fp8e4m3(uintptr_t src, uintptr_t dst, long n, int device,
                              bool sync, int stream_idx) {
  hipStream_t s = get_stream(device, stream_idx);
  int grid = grid_for((n + 7) / 8);
  if ((src | dst) & 15) {
    hipLaunchKernelGGL(cast_fp32_fp8e4m3_scalar_kernel, dim3(grid_for(n)),
                       dim3(256), 0, s, reinterpret_cast<const float*>(src),
                       reinterpret_cast<uint8_t*>(dst), n);
  } else {
    hipLaunchKernelGGL(cast_fp32_fp8e4m3_kernel, dim3(grid), dim3(256), 0, s,
                       reinterpret_cast<const float*>(src),
                       reinterpret_cast<uint8_t*>(dst), n);
  }
  HIP_CHECK(hipGetLastError());
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void cast_fp8e4m3_fp32(uintptr_t src, uintptr_t dst, long n, int device,
                              bool sync, int stream_idx) {
  hipStream_t s = get_stream(device, stream_idx);
  int grid = grid_for((n + 7) / 8);
  if ((src | dst) & 15) {
    hipLaunchKernelGGL(cast_fp8e4m3_fp32_scalar_kernel, dim3(grid_for(n)),
                       dim3(256), 0, s, reinterpret_cast<const uint8_t*>(src),
                       reinterpret_cast<float*>(dst), n);
  } else {
    hipLaunchKernelGGL(cast_fp8e4m3_fp32_kernel, dim3(grid), dim3(256), 0, s,
                       reinterpret_cast<const uint8_t*>(src),
                       reinterpret_cast<float*>(dst), n);
  }
  HIP_CHECK(hipGetLastError());
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void gather_pack(uintptr_t src, uintptr_t dst, int elem_size,
                        std::vector<long> shape, std::vector<long> strides,
                        int device, bool sync) {
  // normalize to 4-D (pad leading dims with 1 / stride 0)
  while (shape.size() < 4) {
    shape.insert(shape.begin(), 1);
    strides.insert(strides.begin(), 0);
  }
  if (shape.size() > 4) throw std::runtime_error("gather_pack: >4D unsupported");
  long n = 1;
  for (long d : shape) n *= d;
  hipStream_t s = get_stream(device);
  int grid = grid_for(n);
  const char* sp = reinterpret_cast<const char*>(src);
  switch (elem_size) {
    case 1:
      hipLaunchKernelGGL((gather_pack_kernel<uint8_t>), dim3(grid), dim3(256), 0,
                         s, sp, reinterpret_cast<uint8_t*>(dst), n, strides[0],
                         strides[1], strides[2], strides[3], shape[0], shape[1],
                         shape[2], shape[3]);
      break;
    case 2:
      hipLaunchKernelGGL((gather_pack_kernel<uint16_t>), dim3(grid), dim3(256),
                         0, s, sp, reinterpret_cast<uint16_t*>(dst), n,
                         strides[0], strides[1], strides[2], strides[3],
                         shape[0], shape[1], shape[2], shape[3]);
      break;
    case 4:
      hipLaunchKernelGGL((gather_pack_kernel<uint32_t>), dim3(grid), dim3(256),
                         0, s, sp, reinterpret_cast<uint32_t*>(dst), n,
                         strides[0], strides[1], strides[2], strides[3],
                         shape[0], shape[1], shape[2], shape[3]);
      break;
    case 8:
      hipLaunchKernelGGL((gather_pack_kernel<uint64_t>), dim3(grid), dim3(256),
                         0, s, sp, reinterpret_cast<uint64_t*>(dst), n,
                         strides[0], strides[1], strides[2], strides[3],
                         shape[0], shape[1], shape[2], shape[3]);
      break;
    default:
      throw std::runtime_error("gather_pack: unsupported element size");
  }
  HIP_CHECK(hipGetLastError());
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

static void rmsnorm_bf16(uintptr_t x, uintptr_t w, uintptr_t out,
                         long rows, int dim, double eps,
                         uintptr_t stream_handle) {
  // launches on the CALLER's stream (torch's current stream) so it
  // composes with torch ops and hipGraph capture
  hipStream_t s = reinterpret_cast<hipStream_t>(stream_handle);
  if (dim % 8 != 0) throw std::runtime_error("rmsnorm: dim % 8 != 0");
  hipLaunchKernelGGL(rmsnorm_bf16_kernel, dim3((uint32_t)rows), dim3(256), 0,
                     s, reinterpret_cast<const uint16_t*>(x),
                     reinterpret_cast<const uint16_t*>(w),
                     reinterpret_cast<uint16_t*>(out), dim, (float)eps);
  HIP_CHECK(hipGetLastError());
}

static void rope_decode_bf16(uintptr_t q, uintptr_t k, uintptr_t cos_tab,
                             uintptr_t sin_tab, uintptr_t pos, int b, int hq,
                             int hk, int d, uintptr_t stream_handle) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream_handle);
  long total = (long)b * (hq + hk) * (d / 2);
  hipLaunchKernelGGL(rope_decode_bf16_kernel, dim3(grid_for(total)),
                     dim3(256), 0, s, reinterpret_cast<uint16_t*>(q),
                     reinterpret_cast<uint16_t*>(k),
                     reinterpret_cast<const float*>(cos_tab),
                     reinterpret_cast<const float*>(sin_tab),
                     reinterpret_cast<const long*>(pos), b, hq, hk, d);
  HIP_CHECK(hipGetLastError());
}

static void image_preprocess(uintptr_t src, uintptr_t dst, int ih, int iw,
                             int oh, int ow, int mode, bool out_bf16,
                             std::vector<float> mean, std::vector<float> stdev,
                             int device, bool sync) {
  if (mean.size() != 3 || stdev.size() != 3)
    throw std::runtime_error("mean/std must have 3 channels");
  hipStream_t s = get_stream(device);
  int grid = grid_for((long)oh * ow);
  hipLaunchKernelGGL(image_preprocess_kernel, dim3(grid), dim3(256), 0, s,
                     reinterpret_cast<const uint8_t*>(src),
                     reinterpret_cast<float*>(dst), ih, iw, oh, ow, mode,
                     out_bf16 ? 1 : 0, mean[0], mean[1], mean[2], stdev[0],
                     stdev[1], stdev[2]);
  HIP_CHECK(hipGetLastError());
  if (sync) {
    py::gil_scoped_release release;
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

PYBIND11_MODULE(_hip_c, m) {
  m.doc() = "client_amd MI355X HIP runtime + CDNA4 kernels";
  m.def("device_count", &device_count);
  m.def("malloc", &hip_malloc, py::arg("device"), py::arg("byte_size"));
  m.def("free", &hip_free, py::arg("ptr"));
  m.def("ipc_get_mem_handle", &ipc_get_mem_handle, py::arg("ptr"));
  m.def("ipc_open_mem_handle", &ipc_open_mem_handle, py::arg("raw_handle"));
  m.def("ipc_close_mem_handle", &ipc_close_mem_handle, py::arg("ptr"));
  m.def("memcpy_h2d", &memcpy_h2d, py::arg("dst"), py::arg("src"),
        py::arg("byte_size"), py::arg("device") = 0, py::arg("sync") = true);
  m.def("memcpy_d2h", &memcpy_d2h, py::arg("src"), py::arg("byte_size"),
        py::arg("device") = 0);
  m.def("memcpy_d2h_into", &memcpy_d2h_into, py::arg("src"), py::arg("dst"),
        py::arg("byte_size"), py::arg("device") = 0);
  m.def("memcpy_d2d", &memcpy_d2d, py::arg("dst"), py::arg("src"),
        py::arg("byte_size"), py::arg("device") = 0, py::arg("sync") = true);
  m.def("stream_sync", &stream_sync, py::arg("device") = 0,
        py::arg("stream_idx") = 0);
  m.def("device_sync", &device_sync);
  m.def("cast_fp32_bf16", &cast_fp32_bf16, py::arg("src"), py::arg("dst"),
        py::arg("n"), py::arg("device") = 0, py::arg("sync") = true,
        py::arg("stream_idx") = 0);
  m.def("cast_fp32_bf16_v2", &cast_fp32_bf16_v2, py::arg("src"),
        py::arg("dst"), py::arg("n"), py::arg("device") = 0,
        py::arg("sync") = true, py::arg("stream_idx") = 0);
  m.def("cast_fp32_bf16_v3", &cast_fp32_bf16_v3, py::arg("src"),
        py::arg("dst"), py::arg("n"), py::arg("device") = 0,
        py::arg("sync") = true, py::arg("stream_idx") = 0);
  m.def("cast_bf16_fp32", &cast_bf16_fp32, py::arg("src"), py::arg("dst"),
        py::arg("n"), py::arg("device") = 0, py::arg("sync") = true,
        py::arg("stream_idx") = 0);
  m.def("cast_fp32_fp8e4m3", &cast_fp32_fp8e4m3, py::arg("src"), py::arg("dst"),
        py::arg("n"), py::arg("device") = 0, py::arg("sync") = true,
        py::arg("stream_idx") = 0);
  m.def("cast_fp8e4m3_fp32", &cast_fp8e4m3_fp32, py::arg("src"), py::arg("dst"),
        py::arg("n"), py::arg("device") = 0, py::arg("sync") = true,
        py::arg("stream_idx") = 0);
  m.def("rmsnorm_bf16", &rmsnorm_bf16, py::arg("x"), py::arg("w"),
        py::arg("out"), py::arg("rows"), py::arg("dim"), py::arg("eps"),
        py::arg("stream_handle"));
  m.def("rope_decode_bf16", &rope_decode_bf16, py::arg("q"), py::arg("k"),
        py::arg("cos_tab"), py::arg("sin_tab"), py::arg("pos"), py::arg("b"),
        py::arg("hq"), py::arg("hk"), py::arg("d"),
        py::arg("stream_handle"));
  m.def("gather_pack", &gather_pack, py::arg("src"), py::arg("dst"),
        py::arg("elem_size"), py::arg("shape"), py::arg("strides"),
        py::arg("device") = 0, py::arg("sync") = true);
  m.def("image_preprocess", &image_preprocess, py::arg("src"), py::arg("dst"),
        py::arg("ih"), py::arg("iw"), py::arg("oh"), py::arg("ow"),
        py::arg("mode") = 0, py::arg("out_bf16") = false,
        py::arg("mean") = std::vector<float>{0.f, 0.f, 0.f},
        py::arg("std") = std::vector<float>{1.f, 1.f, 1.f},
        py::arg("device") = 0, py::arg("sync") = true);
}
