// dtmx CDNA4 (gfx950/MI355X) kernel library — common definitions.
// Hand-written HIP for MI355X only: wave64, MFMA, LDS tiling, glds staging.
// No CUDA compatibility paths, no hipify, no MIOpen.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#define DTMX_WAVE 64
#define DTMX_CUS 256
#define DTMX_XCDS 8

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(8))) _Float16 f16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) short s16x4;

// 16-bit element traits: bf16 and fp16 share every kernel (the reference's
// fp16 multi-precision path); the MFMA intrinsic is the only divergence.
template <typename E>
struct E8;
template <>
struct E8<__bf16> {
  using v8 = bf16x8;
  static __device__ __forceinline__ f32x4 mfma(v8 a, v8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  }
};
template <>
struct E8<_Float16> {
  using v8 = f16x8;
  static __device__ __forceinline__ f32x4 mfma(v8 a, v8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
  }
};

// host-side dispatch over the two 16-bit dtypes
#define DTMX_DISPATCH_16(TYPE, NAME, ...)                         \
  do {                                                            \
    if ((TYPE) == at::kBFloat16) {                                \
      using elem_t = __bf16;                                      \
      __VA_ARGS__                                                 \
    } else if ((TYPE) == at::kHalf) {                             \
      using elem_t = _Float16;                                    \
      __VA_ARGS__                                                 \
    } else {                                                      \
      TORCH_CHECK(false, NAME, ": dtype must be bfloat16/float16"); \
    }                                                             \
  } while (0)

// 16-byte async global->LDS copy (one lane's 16 B; LDS dest is
// wave-uniform base + lane*16 — cdna_hip_programming.md §5).
__device__ __forceinline__ void glds16(const void* gsrc, void* lds_base_wave_uniform) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)gsrc,
      (__attribute__((address_space(3))) void*)lds_base_wave_uniform, 16, 0, 0);
}

__device__ __forceinline__ void wait_vmcnt0() {
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
}

// Fast unsigned division by a runtime invariant divisor (multiply-shift).
// q = (n * magic) >> (32 + shift). Valid for n < 2^31, d >= 1.
struct FastDiv {
  uint32_t d;
  uint32_t magic;   // 0 => power-of-two divisor, use shift only
  uint32_t shift;
  void init(uint32_t d_) {
    d = d_;
    if ((d & (d - 1)) == 0) {  // power of two (incl. 1)
      magic = 0;
      shift = __builtin_ctz(d);
      return;
    }
    // round-up method, s = floor(log2 d): magic = ceil(2^(32+s)/d) < 2^32,
    // exact for all n < 2^32 (Granlund-Montgomery).
    shift = 31 - __builtin_clz(d);
    magic = (uint32_t)(((__uint128_t(1) << (32 + shift)) + d - 1) / d);
  }
  __device__ __forceinline__ uint32_t div(uint32_t n) const {
    if (magic == 0) return n >> shift;
    return (uint32_t)((uint64_t(n) * magic) >> 32 >> shift);
  }
  __device__ __forceinline__ uint32_t mod(uint32_t n, uint32_t q) const {
    return n - q * d;
  }
};

// XCD-aware bijective block remap (cdna_hip_programming.md §5 "XCD swizzle
// must be bijective"): contiguous chunks of the grid per XCD for L2 reuse.
__device__ __forceinline__ uint32_t xcd_swizzle(uint32_t bid, uint32_t nwg) {
  uint32_t q = nwg / DTMX_XCDS, r = nwg % DTMX_XCDS;
  uint32_t xcd = bid % DTMX_XCDS, idx = bid / DTMX_XCDS;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

__device__ __forceinline__ float bf16_to_f32(__bf16 v) { return (float)v; }

__device__ __forceinline__ __bf16 f32_to_bf16(float v) { return (__bf16)v; }

#define DTMX_CHECK(cond, msg) TORCH_CHECK(cond, "dtmx: ", msg)
