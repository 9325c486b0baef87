// 256x128 variant of the 8-phase MFMA GEMM (gemm256.hip) for N=128-class
// shapes — ResNet's 128-channel convs fall below the 256^2 kernel's N>=192
// gate and land on the ~30%-slower 128^2 two-barrier kernel. Geometry:
// BM=256, BN=128, BK=64, 512 threads as 4M x 2N waves (64x64 output per
// wave), 96 KiB LDS (2 buffers x {B, A0, A1} 16 KiB halves). The stream is
// 3 halves per K-tile over 4 phases (phases 0-2 stage, phase 3 none), with
// a 5-half prologue and a CONSTANT vmcnt(4) K-tile boundary (2 halves in
// flight). Fragment schedule per phase q: all 8 B fragments at phase 0
// (live across the tile), one 16-row A fragment per phase; 8 MFMA/phase
// cycling 4 independent accumulators (j), reuse distance 4.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "dtmx_common.h"

namespace dtmx {

template <typename elem_t>
struct Dense256NP {
  using elem = elem_t;
  const elem_t* base;
  const elem_t* zero;
  uint32_t M, K, ld;
};

// row-parity swizzle of the 256^2 kernel (measured best there)
__device__ __forceinline__ uint32_t swz256n(uint32_t byte_off) {
  return byte_off ^ (((byte_off >> 7) & 7) << 4);
}

template <class PA, class PB, typename elem_t>
__launch_bounds__(512, 2) __global__
void gemm256n_kernel(PA pa, PB pb, elem_t* __restrict__ c, uint32_t M,
                     uint32_t N, uint32_t ktiles, uint32_t tiles_n) {
  using V8 = typename E8<elem_t>::v8;
  __shared__ __attribute__((aligned(16))) elem_t smem[2][3][8192];
  const uint32_t t = threadIdx.x;
  const uint32_t wave = t >> 6, lane = t & 63;
  const uint32_t bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const uint32_t bm = (bid / tiles_n) * 256, bn = (bid % tiles_n) * 128;
  const uint32_t wave_m = wave >> 1;  // 0..3: 64-row M strip
  const uint32_t wave_n = wave & 1;   // 0..1: 64-col N strip

  // staging sources: part 0 = B (128 rows), 1 = A0, 2 = A1 (128 rows each);
  // per-lane pointers precomputed with per-tile stride (branch-free loop)
  const elem_t* src0[3][2];
  size_t sstep[3][2];
#pragma unroll
  for (uint32_t part = 0; part < 3; ++part)
#pragma unroll
    for (uint32_t g = 0; g < 2; ++g) {
      uint32_t off = (g * 8 + wave) * 1024 + lane * 16;
      uint32_t lb = swz256n(off);
      uint32_t row = lb >> 7, kb = lb & 127;
      uint32_t k = (kb >> 4) * 8;
      uint32_t m;
      const elem_t* base;
      uint32_t Mlim, ldv;
      if (part == 0) {
        m = bn + row; base = pb.base; Mlim = pb.M; ldv = pb.ld;
      } else {
        m = bm + (part - 1) * 128 + row; base = pa.base; Mlim = pa.M; ldv = pa.ld;
      }
      bool oob = m >= Mlim;
      src0[part][g] = oob ? (part == 0 ? pb.zero : pa.zero)
                          : base + (size_t)m * ldv + k;
      sstep[part][g] = oob ? 0 : 64;
    }
  auto stage_part = [&](auto part_c, uint32_t buf, uint32_t kt) {
    constexpr uint32_t part = decltype(part_c)::value;
#pragma unroll
    for (uint32_t g = 0; g < 2; ++g)
      glds16(src0[part][g] + (size_t)kt * sstep[part][g],
             &smem[buf][part][(g * 8 + wave) * 512]);
  };
  const uint32_t total_halves = ktiles * 3;
  auto stage_stream = [&](uint32_t h) {
    if (h >= total_halves) return;
    uint32_t kt = h / 3, buf = kt & 1;
    switch (h % 3) {
      case 0: stage_part(std::integral_constant<uint32_t, 0>{}, buf, kt); break;
      case 1: stage_part(std::integral_constant<uint32_t, 1>{}, buf, kt); break;
      default: stage_part(std::integral_constant<uint32_t, 2>{}, buf, kt);
    }
  };
  auto wait_vm = [&](uint32_t n) {
    if (n >= 4)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else if (n == 2)
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  };

  // prologue: 5 halves (tile0 complete + tile1 {B, A0})
  uint32_t issued = 0;
  for (; issued < min(5u, total_halves); ++issued) stage_stream(issued);
  wait_vm(2 * (min(5u, total_halves) - min(3u, total_halves)));
  __builtin_amdgcn_s_barrier();

  f32x4 acc[4][4] = {};
  const uint32_t a_slot = 1 + (wave_m >> 1);
  const uint32_t arow0 = (wave_m & 1) * 64;
  const uint32_t wc_local = wave_n * 64;

  V8 bf[8];  // 4 j x 2 kk, live across the tile
  uint32_t S = 0;  // staged-phase counter (3 per tile)
  for (uint32_t kt = 0; kt < ktiles; ++kt) {
    const uint32_t cur = kt & 1;
#pragma unroll
    for (uint32_t q = 0; q < 4; ++q) {
      if (q == 0) {
#pragma unroll
        for (uint32_t j = 0; j < 4; ++j)
#pragma unroll
          for (uint32_t kk = 0; kk < 2; ++kk) {
            uint32_t row = wc_local + j * 16 + (lane & 15);
            uint32_t kbyte = kk * 64 + ((lane >> 4) << 4);
            bf[j * 2 + kk] = *(const V8*)((const char*)&smem[cur][0][0] +
                                          swz256n(row * 128 + kbyte));
          }
      }
      V8 af[2];
#pragma unroll
      for (uint32_t kk = 0; kk < 2; ++kk) {
        uint32_t row = arow0 + q * 16 + (lane & 15);
        uint32_t kbyte = kk * 64 + ((lane >> 4) << 4);
        af[kk] = *(const V8*)((const char*)&smem[cur][a_slot][0] +
                              swz256n(row * 128 + kbyte));
      }
      if (q < 3) {
        stage_stream(5 + S);
        ++S;
      }
      if (q == 0)  // 10-read phase: start draining before the MFMAs
        asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (uint32_t kk = 0; kk < 2; ++kk)
#pragma unroll
        for (uint32_t j = 0; j < 4; ++j)
          acc[q][j] = E8<elem_t>::mfma(af[kk], bf[j * 2 + kk], acc[q][j]);
      __builtin_amdgcn_s_setprio(0);
      if (q == 3) {
        uint32_t issued_now = min(total_halves, 5 + S);
        uint32_t needed = min(total_halves, 3 * (kt + 2));
        wait_vm(2 * (issued_now - needed));
      }
      __builtin_amdgcn_s_barrier();
    }
  }

  // epilogue: direct fragment stores (col = lane&15, row = (lane>>4)*4+r)
  const uint32_t m0 = bm + wave_m * 64, n0 = bn + wc_local;
#pragma unroll
  for (uint32_t i = 0; i < 4; ++i)
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

static const __bf16* zero_page256n(const at::Tensor& like) {
  static at::Tensor z;
  if (!z.defined() || z.device() != like.device())
    z = at::zeros({64}, like.options().dtype(at::kBFloat16));
  return (const __bf16*)z.data_ptr();
}

// y[M,N] = a[M,K] @ b[N,K]^T on the 256x128 kernel (N=128-class probe)
at::Tensor gemm256n_nt(const at::Tensor& a, const at::Tensor& b) {
  DTMX_DISPATCH_16(a.scalar_type(), "gemm256n_nt", {
    TORCH_CHECK(a.is_cuda() && b.is_cuda() && a.dim() == 2 && b.dim() == 2 &&
                    a.size(1) == b.size(1),
                "gemm256n_nt: a[M,K], b[N,K] CUDA 16-bit");
    auto ac = a.contiguous();
    auto bc = b.contiguous();
    uint32_t M = ac.size(0), K = ac.size(1), N = bc.size(0);
    TORCH_CHECK(K % 64 == 0, "gemm256n_nt: K % 64 != 0");
    auto y = at::empty({(long)M, (long)N}, a.options());
    Dense256NP<elem_t> pa{(const elem_t*)ac.data_ptr(),
                          (const elem_t*)zero_page256n(a), M, K, K};
    Dense256NP<elem_t> pb{(const elem_t*)bc.data_ptr(),
                          (const elem_t*)zero_page256n(a), N, K, K};
    uint32_t ktiles = (K + 63) / 64;
    uint32_t tiles_m = (M + 255) / 256, tiles_n = (N + 127) / 128;
    dim3 grid(tiles_m * tiles_n);
    hipStream_t s = at::hip::getCurrentHIPStream().stream();
    gemm256n_kernel<Dense256NP<elem_t>, Dense256NP<elem_t>, elem_t>
        <<<grid, 512, 0, s>>>(pa, pb, (elem_t*)y.data_ptr(), M, N, ktiles,
                              tiles_n);
    return y;
  });
  return at::Tensor();
}

}  // namespace dtmx
