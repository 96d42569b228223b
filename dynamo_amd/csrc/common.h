// Common device helpers for dynamo_amd HIP kernels (gfx950 / CDNA4 only).
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  - wavefront = 64 lanes; block sizes are multiples of 64
//  - bf16 traffic is always vectorized as short4/short8 (scalar bf16 loads
//    are ~2-2.5x slower; guide Common-mistake #2)
//  - fp32 accumulation everywhere
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <stdint.h>

#define WAVE_SIZE 64

#define DEVINL __device__ __forceinline__

// ---- vector types -------------------------------------------------------
// 8 bf16 = 16 B, the coalescing sweet spot.
typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(4))) float float4v;
typedef __attribute__((ext_vector_type(2))) float float2v;

// MFMA fragment types for mfma_f32_16x16x32_bf16 (gfx950).
typedef __attribute__((ext_vector_type(8))) short bf16x8;   // A/B operand: 8 bf16
typedef __attribute__((ext_vector_type(4))) float f32x4;    // C/D accum: 4 fp32
// MFMA fragment types for mfma_f32_32x32x16_bf16.
typedef __attribute__((ext_vector_type(16))) float f32x16;  // C/D accum: 16 fp32

DEVINL float bf16_to_f32(short u) {
  union { float f; uint32_t i; } v;
  v.i = ((uint32_t)(uint16_t)u) << 16;
  return v.f;
}

DEVINL short f32_to_bf16(float f) {
  // round-to-nearest-even
  union { float f; uint32_t i; } v;
  v.f = f;
  uint32_t lsb = (v.i >> 16) & 1;
  uint32_t rounded = v.i + 0x7fff + lsb;
  return (short)(rounded >> 16);
}

// ---- wave reductions ----------------------------------------------------
DEVINL float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE_SIZE);
  return x;
}

DEVINL float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, WAVE_SIZE));
  return x;
}

// Reduce across a contiguous group of `width` lanes (width power of 2).
template <int WIDTH>
DEVINL float group_reduce_sum(float x) {
#pragma unroll
  for (int off = WIDTH / 2; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE_SIZE);
  return x;
}

// ---- block reduction (needs extern LDS scratch of >= num_waves floats) --
// Block-level sum using a caller-provided LDS buffer (size: block/64 floats).
DEVINL float block_reduce_sum(float x, float* lds_scratch) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  const int nwaves = blockDim.x / WAVE_SIZE;
  x = wave_reduce_sum(x);
  if (lane == 0) lds_scratch[wid] = x;
  __syncthreads();
  float total = 0.f;
  if (threadIdx.x < (unsigned)nwaves) total = lds_scratch[threadIdx.x];
  if (wid == 0) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) total += __shfl_xor(total, off, WAVE_SIZE);
    if (lane == 0) lds_scratch[0] = total;
  }
  __syncthreads();
  return lds_scratch[0];
}

// ---- misc ---------------------------------------------------------------
DEVINL int64_t cdiv64(int64_t a, int64_t b) { return (a + b - 1) / b; }

// ---- fp8 (OCP e4m3) KV cache support --------------------------------------
typedef __attribute__((ext_vector_type(8))) unsigned char uchar8;

#if !defined(__HIP_DEVICE_COMPILE__)
// host pass only parses kernel bodies; device builtins are unavailable
DEVINL bf16x8 fp8x8_to_bf16x8(uchar8) { return bf16x8{}; }
#else
// 8 packed e4m3 bytes -> 8 bf16 (MFMA A/B fragment shape): 4x native
// v_cvt_pk_f32_fp8 + 4x v_cvt_pk_bf16_f32
DEVINL bf16x8 fp8x8_to_bf16x8(uchar8 v) {
  union { uchar8 u8; unsigned int w[2]; } in;
  in.u8 = v;
  unsigned int out[4];
  // word-select operand must be a compile-time constant
  float2v f0 = __builtin_amdgcn_cvt_pk_f32_fp8((int)in.w[0], false);
  float2v f1 = __builtin_amdgcn_cvt_pk_f32_fp8((int)in.w[0], true);
  float2v f2 = __builtin_amdgcn_cvt_pk_f32_fp8((int)in.w[1], false);
  float2v f3 = __builtin_amdgcn_cvt_pk_f32_fp8((int)in.w[1], true);
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(out[0]) : "v"(f0.x), "v"(f0.y));
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(out[1]) : "v"(f1.x), "v"(f1.y));
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(out[2]) : "v"(f2.x), "v"(f2.y));
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(out[3]) : "v"(f3.x), "v"(f3.y));
  return *reinterpret_cast<bf16x8*>(out);
}
#endif


// 16-lane (DPP row) reductions via row_ror rotations: pure VALU, no
// ds_bpermute — cross-lane LDS-unit shuffles have ~60-cycle latency and
// serialize softmax dependency chains (measured: replacing them in the
// prefill kernel's P exchange took it 260 -> 569 TF).
template <int CTRL>
DEVINL float dpp_movf(float x) {
  int i = __builtin_bit_cast(int, x);
  i = __builtin_amdgcn_update_dpp(0, i, CTRL, 0xf, 0xf, false);
  return __builtin_bit_cast(float, i);
}

// max over the 16 lanes of a DPP row, broadcast to every lane of the row
DEVINL float row16_reduce_max(float x) {
  x = fmaxf(x, dpp_movf<0x128>(x));   // row_ror:8
  x = fmaxf(x, dpp_movf<0x124>(x));   // row_ror:4
  x = fmaxf(x, dpp_movf<0x122>(x));   // row_ror:2
  x = fmaxf(x, dpp_movf<0x121>(x));   // row_ror:1
  return x;
}

DEVINL float row16_reduce_sum(float x) {
  x += dpp_movf<0x128>(x);
  x += dpp_movf<0x124>(x);
  x += dpp_movf<0x122>(x);
  x += dpp_movf<0x121>(x);
  return x;
}

// 8-lane-group reductions via XOR-mask butterflies in DPP: quad_perm
// xor1 (0xB1) + xor2 (0x4E) + row_half_mirror (= xor7 within the 8-lane
// half, 0x141). Masks {1,2,7} span bits 0-2, so every lane ends with the
// full 8-group value. Pure VALU — no ds_bpermute.
DEVINL float group8_reduce_sum(float x) {
  x += dpp_movf<0xB1>(x);
  x += dpp_movf<0x4E>(x);
  x += dpp_movf<0x141>(x);
  return x;
}

DEVINL float group8_reduce_max(float x) {
  x = fmaxf(x, dpp_movf<0xB1>(x));
  x = fmaxf(x, dpp_movf<0x4E>(x));
  x = fmaxf(x, dpp_movf<0x141>(x));
  return x;
}

#define HIP_CHECK_KERNEL()                                                \
  do {                                                                    \
    hipError_t _e = hipGetLastError();                                    \
    if (_e != hipSuccess) {                                               \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(_e)); \
    }                                                                     \
  } while (0)
