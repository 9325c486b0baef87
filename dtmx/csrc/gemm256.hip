// 256x256 8-phase MFMA GEMM for gfx950 (cdna_hip_programming.md §5 "The
// 256² 8-phase template"): the deep-pipelined structure past the 128²-tile
// two-barrier ceiling (~900 TF). Geometry: BM=BN=256, BK=64, 512 threads
// (8 waves as 2M x 4N, 128x64 output per wave), 128 KiB LDS (2 buffers x
// {B0,B1,A0,A1} 16 KiB half-slots), st_16x32 XOR swizzle on the glds SOURCE
// address (rule 21), counted vmcnt(6) at K-tile boundaries only.
//
// Schedule (re-derived from the template's constraints): per K-tile, phase 0
// LDS-reads ALL B fragments (8 x ds_read_b128 -> registers live across the
// tile) plus the first A quarter (4 reads); phases 1-3 read one A quarter
// each; every phase stages ONE half-tile of the stream (order per tile:
// B0,B1,A0,A1, offset so tile t's phases stage {A1(t+1), B0(t+2), B1(t+2),
// A0(t+2)}); the 7-half prologue + boundary vmcnt(6) then guarantee every
// half of tile t+1 is globally visible at its phase 0 while 3 halves of
// t+2 stay in flight. Half-slots recycle mod 2 tiles: B slots are free
// after their tile's phase 0, A slots free quarter-by-quarter ahead of the
// ~900-cycle HBM round trip of the overwriting glds.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "dtmx_common.h"

namespace dtmx {

template <typename elem_t>
struct Dense256P {
  using elem = elem_t;
  const elem_t* base;
  const elem_t* zero;
  uint32_t M, K, ld;
  __device__ __forceinline__ const void* addr(uint32_t m, uint32_t k8) const {
    if (m >= M || k8 * 8 >= K) return zero;
    return base + (size_t)m * ld + k8 * 8;
  }
};

// swizzle within a 16 KiB half image ([128 rows][128 B]). Default: the
// row-parity XOR of the proven 128^2 kernel (byte ^= (row&7)<<4 — measured
// 0.17% bank-conflict cycles there vs 2.5% for st_16x32 on this layout);
// DTMX_G256_ST16 compiles the guide's st_16x32 for A/B.
template <int ST16>
__device__ __forceinline__ uint32_t swz256(uint32_t byte_off) {
  if constexpr (ST16) return byte_off ^ (((byte_off >> 9) & 1) << 5);
  return byte_off ^ (((byte_off >> 7) & 7) << 4);
}

template <class PA, class PB, typename elem_t, int NOBAR1 = 0, int ST16 = 0>
__launch_bounds__(512, 2) __global__
void gemm256_kernel(PA pa, PB pb, elem_t* __restrict__ c, uint32_t M,
                    uint32_t N, uint32_t ktiles, uint32_t tiles_n) {
  using V8 = typename E8<elem_t>::v8;
  __shared__ __attribute__((aligned(16))) elem_t smem[2][4][8192];
  const uint32_t t = threadIdx.x;
  const uint32_t wave = t >> 6, lane = t & 63;
  const uint32_t bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const uint32_t bm = (bid / tiles_n) * 256, bn = (bid % tiles_n) * 256;
  const uint32_t am_half = wave >> 2;          // this wave's A half (0/1)
  const uint32_t wc = (wave & 3) * 64;         // this wave's B column base

  // stage one 16 KiB half: part 0/1 = B halves, 2/3 = A halves. 2 glds per
  // wave; LDS dest lane-linear, source address carries the st_16x32 swizzle.
  // Per-lane source pointers are PRECOMPUTED with a per-tile stride (0 for
  // out-of-range rows, which then re-read the zero page every tile) — the
  // in-loop stage is two pointer adds + two glds, no branches (the addr()
  // select/branch form cost ~35% of the kernel in exec-mask dances).
  const elem_t* src0[4][2];
  size_t sstep[4][2];
#pragma unroll
  for (uint32_t part = 0; part < 4; ++part)
#pragma unroll
    for (uint32_t g = 0; g < 2; ++g) {
      uint32_t off = (g * 8 + wave) * 1024 + lane * 16;  // bytes in half
      uint32_t lb = swz256<ST16>(off);
      uint32_t row = lb >> 7, kb = lb & 127;
      uint32_t k8 = kb >> 4;
      const void* p = part < 2 ? pb.addr(bn + part * 128 + row, k8)
                               : pa.addr(bm + (part - 2) * 128 + row, k8);
      bool oob = part < 2 ? (bn + part * 128 + row >= pb.M)
                          : (bm + (part - 2) * 128 + row >= pa.M);
      src0[part][g] = (const elem_t*)p;
      sstep[part][g] = oob ? 0 : 64;  // K%64==0 enforced by the host
    }
  // part must be a compile-time constant: runtime indexing of src0 would
  // send the pointer array to scratch (rule 20)
  auto stage_part = [&](auto part_c, uint32_t buf, uint32_t kt) {
    constexpr uint32_t part = decltype(part_c)::value;
#pragma unroll
    for (uint32_t g = 0; g < 2; ++g)
      glds16(src0[part][g] + (size_t)kt * sstep[part][g],
             &smem[buf][part][(g * 8 + wave) * 512]);
  };
  // global half stream: index h -> tile h/4, part order {B0,B1,A0,A1}
  const uint32_t total_halves = ktiles * 4;
  auto stage_stream = [&](uint32_t h) {
    if (h >= total_halves) return;
    uint32_t kt = h >> 2, buf = kt & 1;
    switch (h & 3) {
      case 0: stage_part(std::integral_constant<uint32_t, 0>{}, buf, kt); break;
      case 1: stage_part(std::integral_constant<uint32_t, 1>{}, buf, kt); break;
      case 2: stage_part(std::integral_constant<uint32_t, 2>{}, buf, kt); break;
      default: stage_part(std::integral_constant<uint32_t, 3>{}, buf, kt);
    }
  };
  auto wait_vm = [&](uint32_t n) {  // immediate-operand dispatch
    if (n >= 6)
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else if (n == 4)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else if (n == 2)
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  };

  // prologue: 7 halves of the stream (tile0 complete + tile1 B0,B1,A0)
  uint32_t issued = 0;
  for (; issued < min(4u, total_halves); ++issued) stage_stream(issued);
  wait_vm(total_halves > 4 ? 4 : 0);
  for (; issued < min(7u, total_halves); ++issued) stage_stream(issued);
  // tile 0 landed; up to 3 halves in flight
  wait_vm(2 * (issued - min(4u, total_halves)));
  __builtin_amdgcn_s_barrier();

  f32x4 acc[8][4] = {};
  // NOTE: address smem[...] DIRECTLY in each read — stashing the base in a
  // runtime-indexed pointer array erases the addrspace(3) provenance and
  // hipcc lowers the reads as FLAT loads (VM-counted: the glds pipeline then
  // drains with vmcnt(0) before every MFMA cluster; measured 427 TF).
  const uint32_t a_slot = 2 + am_half;
  const uint32_t b_slot = wc >> 7;
  const uint32_t wc_local = wc & 127;

  V8 bf[8];  // 4 j x 2 kk, live across the tile's four phases
  uint32_t P = 0;
  for (uint32_t kt = 0; kt < ktiles; ++kt) {
    const uint32_t cur = kt & 1;
#pragma unroll
    for (uint32_t q = 0; q < 4; ++q, ++P) {
      if (q == 0) {
#pragma unroll
        for (uint32_t j = 0; j < 4; ++j)
#pragma unroll
          for (uint32_t kk = 0; kk < 2; ++kk) {
            uint32_t row = wc_local + j * 16 + (lane & 15);
            uint32_t kbyte = kk * 64 + ((lane >> 4) << 4);
            bf[j * 2 + kk] = *(const V8*)((const char*)&smem[cur][b_slot][0] +
                                          swz256<ST16>(row * 128 + kbyte));
          }
      }
      V8 af[2][2];  // i in {2q, 2q+1} x kk
#pragma unroll
      for (uint32_t ii = 0; ii < 2; ++ii)
#pragma unroll
        for (uint32_t kk = 0; kk < 2; ++kk) {
          uint32_t row = (2 * q + ii) * 16 + (lane & 15);
          uint32_t kbyte = kk * 64 + ((lane >> 4) << 4);
          af[ii][kk] = *(const V8*)((const char*)&smem[cur][a_slot][0] +
                                    swz256<ST16>(row * 128 + kbyte));
        }
      stage_stream(7 + P);
      if (q == 0)  // 12-read phase: start draining before the barrier
        asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
      if constexpr (!NOBAR1) __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
      // kk OUTER of (ii,j): 8 independent accumulators between reuses of
      // any acc — kk-inner order made every MFMA pair dependent
#pragma unroll
      for (uint32_t kk = 0; kk < 2; ++kk)
#pragma unroll
        for (uint32_t ii = 0; ii < 2; ++ii)
#pragma unroll
          for (uint32_t j = 0; j < 4; ++j)
            acc[2 * q + ii][j] = E8<elem_t>::mfma(af[ii][kk], bf[j * 2 + kk],
                                                  acc[2 * q + ii][j]);
      __builtin_amdgcn_s_setprio(0);
      if (q == 3) {
        // K-tile boundary: everything of tile kt+1 must be visible; allow
        // the newest in-flight halves (tile kt+2's) to keep flying
        uint32_t issued_now = min(total_halves, 7 + P + 1);
        uint32_t needed = min(total_halves, 4 * (kt + 2));
        wait_vm(2 * (issued_now - needed));
      }
      __builtin_amdgcn_s_barrier();
    }
  }

  // epilogue: direct fragment stores (col = lane&15, row = (lane>>4)*4+r)
  const uint32_t m0 = bm + am_half * 128, n0 = bn + wc;
#pragma unroll
  for (uint32_t i = 0; i < 8; ++i)
#pragma unroll
    for (uint32_t j = 0; j < 4; ++j) {
      uint32_t n = n0 + j * 16 + (lane & 15);
      if (n >= N) continue;
#pragma unroll
      for (uint32_t r = 0; r < 4; ++r) {
        uint32_t m = m0 + i * 16 + ((lane >> 4) << 2) + r;
        if (m >= M) continue;
        c[(size_t)m * N + n] = (elem_t)acc[i][j][r];
      }
    }
}

// ---- 32x32x16 MFMA variant (probe): same schedule, bigger matrix op
// (2382 vs 2075 TF ubench ceiling). Per-wave 128x64 = 4 M-tiles x 2
// N-tiles of 32^2; phase q computes M-tile q (8 MFMA/phase).
typedef __attribute__((ext_vector_type(16))) float f32x16;

template <typename elem_t>
__device__ __forceinline__ f32x16 mfma32(typename E8<elem_t>::v8 a,
                                         typename E8<elem_t>::v8 b, f32x16 c) {
  if constexpr (std::is_same_v<elem_t, __bf16>)
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  else
    return __builtin_amdgcn_mfma_f32_32x32x16_f16(a, b, c, 0, 0, 0);
}

template <class PA, class PB, typename elem_t>
__launch_bounds__(512, 2) __global__
void gemm256_m32_kernel(PA pa, PB pb, elem_t* __restrict__ c, uint32_t M,
                        uint32_t N, uint32_t ktiles, uint32_t tiles_n) {
  using V8 = typename E8<elem_t>::v8;
  __shared__ __attribute__((aligned(16))) elem_t smem[2][4][8192];
  const uint32_t t = threadIdx.x;
  const uint32_t wave = t >> 6, lane = t & 63;
  const uint32_t bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const uint32_t bm = (bid / tiles_n) * 256, bn = (bid % tiles_n) * 256;
  const uint32_t am_half = wave >> 2;
  const uint32_t wc = (wave & 3) * 64;

  const elem_t* src0[4][2];
  size_t sstep[4][2];
#pragma unroll
  for (uint32_t part = 0; part < 4; ++part)
#pragma unroll
    for (uint32_t g = 0; g < 2; ++g) {
      uint32_t off = (g * 8 + wave) * 1024 + lane * 16;
      uint32_t lb = swz256<0>(off);
      uint32_t row = lb >> 7, kb = lb & 127;
      uint32_t k = (kb >> 4) * 8;
      if (part < 2) {
        uint32_t m = bn + part * 128 + row;
        bool oob = m >= pb.M;
        src0[part][g] = oob ? pb.zero : pb.base + (size_t)m * pb.ld + k;
        sstep[part][g] = oob ? 0 : 64;
      } else {
        uint32_t m = bm + (part - 2) * 128 + row;
        bool oob = m >= pa.M;
        src0[part][g] = oob ? pa.zero : pa.base + (size_t)m * pa.ld + k;
        sstep[part][g] = oob ? 0 : 64;
      }
    }
  auto stage_part = [&](auto part_c, uint32_t buf, uint32_t kt) {
    constexpr uint32_t part = decltype(part_c)::value;
#pragma unroll
    for (uint32_t g = 0; g < 2; ++g)
      glds16(src0[part][g] + (size_t)kt * sstep[part][g],
             &smem[buf][part][(g * 8 + wave) * 512]);
  };
  const uint32_t total_halves = ktiles * 4;
  auto stage_stream = [&](uint32_t h) {
    if (h >= total_halves) return;
    uint32_t kt = h >> 2, buf = kt & 1;
    switch (h & 3) {
      case 0: stage_part(std::integral_constant<uint32_t, 0>{}, buf, kt); break;
      case 1: stage_part(std::integral_constant<uint32_t, 1>{}, buf, kt); break;
      case 2: stage_part(std::integral_constant<uint32_t, 2>{}, buf, kt); break;
      default: stage_part(std::integral_constant<uint32_t, 3>{}, buf, kt);
    }
  };
  auto wait_vm = [&](uint32_t n) {
    if (n >= 6)
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else if (n == 4)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else if (n == 2)
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  };

  uint32_t issued = 0;
  for (; issued < min(4u, total_halves); ++issued) stage_stream(issued);
  wait_vm(total_halves > 4 ? 4 : 0);
  for (; issued < min(7u, total_halves); ++issued) stage_stream(issued);
  wait_vm(2 * (issued - min(4u, total_halves)));
  __builtin_amdgcn_s_barrier();

  f32x16 acc4[4][2] = {};
  const uint32_t a_slot = 2 + am_half;
  const uint32_t b_slot = wc >> 7;
  const uint32_t wc_local = wc & 127;

  V8 bf[8];  // 2 j x 4 kk
  uint32_t P = 0;
  for (uint32_t kt = 0; kt < ktiles; ++kt) {
    const uint32_t cur = kt & 1;
#pragma unroll
    for (uint32_t q = 0; q < 4; ++q, ++P) {
      if (q == 0) {
#pragma unroll
        for (uint32_t j = 0; j < 2; ++j)
#pragma unroll
          for (uint32_t kk = 0; kk < 4; ++kk) {
            uint32_t row = wc_local + j * 32 + (lane & 31);
            uint32_t kbyte = kk * 32 + ((lane >> 5) << 4);
            bf[j * 4 + kk] = *(const V8*)((const char*)&smem[cur][b_slot][0] +
                                          swz256<0>(row * 128 + kbyte));
          }
      }
      V8 af[4];
#pragma unroll
      for (uint32_t kk = 0; kk < 4; ++kk) {
        uint32_t row = q * 32 + (lane & 31);
        uint32_t kbyte = kk * 32 + ((lane >> 5) << 4);
        af[kk] = *(const V8*)((const char*)&smem[cur][a_slot][0] +
                              swz256<0>(row * 128 + kbyte));
      }
      stage_stream(7 + P);
      if (q == 0)
        asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (uint32_t kk = 0; kk < 4; ++kk)
#pragma unroll
        for (uint32_t j = 0; j < 2; ++j)
          acc4[q][j] = mfma32<elem_t>(af[kk], bf[j * 4 + kk], acc4[q][j]);
      __builtin_amdgcn_s_setprio(0);
      if (q == 3) {
        uint32_t issued_now = min(total_halves, 7 + P + 1);
        uint32_t needed = min(total_halves, 4 * (kt + 2));
        wait_vm(2 * (issued_now - needed));
      }
      __builtin_amdgcn_s_barrier();
    }
  }

  // epilogue: 32x32 C map col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)
  const uint32_t m0 = bm + am_half * 128, n0 = bn + wc;
#pragma unroll
  for (uint32_t i = 0; i < 4; ++i)
#pragma unroll
    for (uint32_t j = 0; j < 2; ++j) {
      uint32_t n = n0 + j * 32 + (lane & 31);
      if (n >= N) continue;
#pragma unroll
      for (uint32_t r = 0; r < 16; ++r) {
        uint32_t m = m0 + i * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
        if (m >= M) continue;
        c[(size_t)m * N + n] = (elem_t)acc4[i][j][r];
      }
    }
}

static const __bf16* zero_page256(const at::Tensor& like) {
  static at::Tensor z;
  if (!z.defined() || z.device() != like.device())
    z = at::zeros({64}, like.options().dtype(at::kBFloat16));
  return (const __bf16*)z.data_ptr();
}

// y[M,N] = a[M,K] @ b[N,K]^T  (the linear_fwd contraction) on the 2562
// 8-phase kernel — the big-dense-shape path.
at::Tensor gemm256_nt(const at::Tensor& a, const at::Tensor& b) {
  DTMX_DISPATCH_16(a.scalar_type(), "gemm256_nt", {
    TORCH_CHECK(a.is_cuda() && b.is_cuda() && a.dim() == 2 && b.dim() == 2 &&
                    a.size(1) == b.size(1),
                "gemm256_nt: a[M,K], b[N,K] CUDA 16-bit");
    auto ac = a.contiguous();
    auto bc = b.contiguous();
    uint32_t M = ac.size(0), K = ac.size(1), N = bc.size(0);
    TORCH_CHECK(K % 64 == 0, "gemm256_nt: K % 64 != 0 (precomputed-stride "
                             "staging; route other shapes to the 128^2 kernel)");
    auto y = at::empty({(long)M, (long)N}, a.options());
    Dense256P<elem_t> pa{(const elem_t*)ac.data_ptr(),
                         (const elem_t*)zero_page256(a), M, K, K};
    Dense256P<elem_t> pb{(const elem_t*)bc.data_ptr(),
                         (const elem_t*)zero_page256(a), N, K, K};
    uint32_t ktiles = (K + 63) / 64;
    uint32_t tiles_m = (M + 255) / 256, tiles_n = (N + 255) / 256;
    dim3 grid(tiles_m * tiles_n);
    hipStream_t s = at::hip::getCurrentHIPStream().stream();
    // phase-aligned waves without the pre-MFMA barrier measured +7%
    // (1126 vs 1048 TF @8192^3); DTMX_G256_BAR1=1 restores the two-barrier
    // phase for A/B
    static const int variant = [] {
      const char* b = getenv("DTMX_G256_BAR1");
      const char* w = getenv("DTMX_G256_ST16");
      const char* m = getenv("DTMX_G256_M32");
      if (m && m[0] == '1') return 4;
      return (b && b[0] == '1' ? 0 : 1) | (w && w[0] == '1' ? 2 : 0);
    }();
    switch (variant) {
      case 4:  // 32x32x16 MFMA probe (same stream schedule, f32x16 acc)
        gemm256_m32_kernel<Dense256P<elem_t>, Dense256P<elem_t>, elem_t>
            <<<grid, 512, 0, s>>>(pa, pb, (elem_t*)y.data_ptr(), M, N, ktiles, tiles_n);
        break;
      case 0:
        gemm256_kernel<Dense256P<elem_t>, Dense256P<elem_t>, elem_t, 0, 0>
            <<<grid, 512, 0, s>>>(pa, pb, (elem_t*)y.data_ptr(), M, N, ktiles, tiles_n);
        break;
      case 1:
        gemm256_kernel<Dense256P<elem_t>, Dense256P<elem_t>, elem_t, 1, 0>
            <<<grid, 512, 0, s>>>(pa, pb, (elem_t*)y.data_ptr(), M, N, ktiles, tiles_n);
        break;
      case 2:
        gemm256_kernel<Dense256P<elem_t>, Dense256P<elem_t>, elem_t, 0, 1>
            <<<grid, 512, 0, s>>>(pa, pb, (elem_t*)y.data_ptr(), M, N, ktiles, tiles_n);
        break;
      default:
        gemm256_kernel<Dense256P<elem_t>, Dense256P<elem_t>, elem_t, 1, 1>
            <<<grid, 512, 0, s>>>(pa, pb, (elem_t*)y.data_ptr(), M, N, ktiles, tiles_n);
    }
    return y;
  });
  return at::Tensor();
}

}  // namespace dtmx
