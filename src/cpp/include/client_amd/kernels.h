// C ABI for the CDNA4 kernels (implemented in
// client_amd/ops/csrc/kernels.hip, one compilation unit shared with the
// Python extension). Compile your program with hipcc and add
// kernels.hip to its sources, or link the cmake `clientamd_kernels`
// target. Every launcher takes an explicit hipStream_t and returns
// hipGetLastError() — callers own synchronization.
#pragma once

#include <hip/hip_runtime_api.h>

extern "C" {

// fp32 -> bf16 wire-exact truncation pack (the tritonclient BF16 codec
// semantics, 5.2 TB/s measured on MI355X)
hipError_t ca_cast_fp32_bf16(const void* src, void* dst, long n,
                             hipStream_t stream);
// bf16 -> fp32 zero-extension unpack
hipError_t ca_cast_bf16_fp32(const void* src, void* dst, long n,
                             hipStream_t stream);
// fp32 <-> fp8 OCP e4m3fn (RNE)
hipError_t ca_cast_fp32_fp8e4m3(const void* src, void* dst, long n,
                                hipStream_t stream);
hipError_t ca_cast_fp8e4m3_fp32(const void* src, void* dst, long n,
                                hipStream_t stream);
// strided (<=4-D) -> contiguous gather; shape/strides are 4-long host
// arrays in elements
hipError_t ca_gather_pack(const void* src, void* dst, int elem_size,
                          const long* shape, const long* strides,
                          hipStream_t stream);
// u8 HWC -> bilinear resize -> normalize -> CHW fp32/bf16
// (mode 0 = (x-mean)*std, 1 = INCEPTION, 2 = VGG)
hipError_t ca_image_preprocess(const void* src, void* dst, int ih, int iw,
                               int oh, int ow, int mode, int out_bf16,
                               const float* mean, const float* stdev,
                               hipStream_t stream);
// fused bf16 RMSNorm / per-row-position decode RoPE (LLM serving)
hipError_t ca_rmsnorm_bf16(const void* x, const void* w, void* out, long rows,
                           int dim, float eps, hipStream_t stream);
hipError_t ca_rope_decode_bf16(void* q, void* k, const void* cos_tab,
                               const void* sin_tab, const void* pos, int b,
                               int hq, int hk, int d, hipStream_t stream);

}  // extern "C"
