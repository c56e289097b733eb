// Fused dense-MLP kernels for gfx950 (CDNA4): bf16 MFMA GEMM with fused
// bias + ReLU epilogues, plus backward (dgrad / wgrad / bias-grad) kernels.
// Replaces the reference's dense path (torch Linear over hipBLASLt + an
// elementwise cascade) with MFMA/LDS-tiled kernels (BASELINE north star:
// "dense-side fused MLP ... hand-written CDNA4 HIP kernels").
//
// Structure per the CDNA HIP guide §5: 128x128 C-tile, BK=32 K-step,
// 4 waves x 64x64 per-wave tiles of mfma_f32_16x16x32_bf16 fragments,
// double-buffered LDS.  Both operands are staged K-CONTIGUOUS ([row][k],
// +8-short row pad -> conflict-free ds_read_b128 fragment loads): the GEMM
// is the NT form C[M,N] = A[M,K] @ B[N,K]^T, which consumes torch Linear
// weights ([out,in]) without any transpose in the forward.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

using bf16 = __hip_bfloat16;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) short;

namespace {

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int PAD = 8;                 // row pad (shorts): banks coprime
constexpr int LDK = BK + PAD;          // LDS row stride

__device__ __forceinline__ float bf2f(short x) {
  union { float f; unsigned u; } v;
  v.u = ((unsigned)(unsigned short)x) << 16;
  return v.f;
}

__device__ __forceinline__ short f2bf(float f) {
  union { float f; unsigned u; } v;
  v.f = f;
  unsigned lsb = (v.u >> 16) & 1;
  v.u += 0x7fff + lsb;  // round-to-nearest-even
  return (short)(v.u >> 16);
}

// ---------------------------------------------------------------------------
// NT GEMM: C[M,N] = act(A[M,K] @ B[N,K]^T + bias[N]); A,B bf16 row-major.
// mfma_f32_16x16x32_bf16 fragment maps (verified numerically on gfx950):
//   A frag: lane holds A[i = lane&15][k = (lane>>4)*8 + j]
//   B frag: lane holds B^T[k][n = lane&15] = B[n][k], k = (lane>>4)*8 + j
//   C frag: lane holds C[row = (lane>>4)*4 + r][col = lane&15]
// Both fragments read 8 contiguous shorts from a [row][k] LDS image.
// ---------------------------------------------------------------------------
// TBN: output tile N-width (128, or 64 for skinny layers so the grid fills
// 256 CUs).  TBK: K-step (32 or 64 — 64 halves the barrier count).
// TRANS_B: second operand given as [Kdim, N] row-major (the dgrad case
// dX = g @ W with W [N,K]) — staged through a transposed LDS image (scalar
// writes) so fragment reads stay contiguous.
// T14 async-STAGE split (guide §6 G15): the next tile's global loads are
// issued into REGISTERS before the MFMA block (HBM latency hides under the
// compute), LDS writes happen after it.
template <int ACT, bool STORE_F32, bool TRANS_B, int TBN, int TBK>
__global__ __launch_bounds__(256) void gemm_nt_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    const float* __restrict__ bias, void* __restrict__ C, int M, int N,
    int K) {
  constexpr int LK = TBK + PAD;
  __shared__ short lds_a[2][BM * LK];
  __shared__ short lds_b[2][TBN * LK];
  constexpr int WN = TBN / 2;  // per-wave N extent
  constexpr int NFRAG = WN / 16;
  constexpr int PA = BM * TBK / (256 * 8);   // A pieces per thread
  constexpr int PB = TBN * TBK / (256 * 8);  // B pieces per thread
  constexpr int ACPR = TBK / 8;              // A pieces per row

  const int n_tiles_n = (N + TBN - 1) / TBN;
  const int m0 = (blockIdx.x / n_tiles_n) * BM;
  const int n0 = (blockIdx.x % n_tiles_n) * TBN;

  const int tid = threadIdx.x;
  const int lane = tid % 64;
  const int wave = tid / 64;
  const int wr = wave / 2, wc = wave % 2;  // 64 x WN wave tile position
  const int fi = lane & 15;
  const int fk8 = (lane >> 4) * 8;

  f32x4 acc[4][NFRAG] = {};

  auto stage_load = [&](int k0, bf16x8 (&ra)[PA], bf16x8 (&rb)[PB]) {
#pragma unroll
    for (int i = 0; i < PA; ++i) {
      const int p = tid + i * 256;
      const int r = p / ACPR, c8 = (p % ACPR) * 8;
      ra[i] = bf16x8{};
      if (m0 + r < M)
        ra[i] = *(const bf16x8*)(A + (int64_t)(m0 + r) * K + k0 + c8);
    }
#pragma unroll
    for (int i = 0; i < PB; ++i) {
      const int p = tid + i * 256;
      rb[i] = bf16x8{};
      if (!TRANS_B) {
        const int r = p / ACPR, c8 = (p % ACPR) * 8;  // image [TBN][TBK]
        if (n0 + r < N)
          rb[i] = *(const bf16x8*)(B + (int64_t)(n0 + r) * K + k0 + c8);
      } else {
        const int r = p / (TBN / 8), c8 = (p % (TBN / 8)) * 8;  // mem [TBK][N]
        if (n0 + c8 < N)
          rb[i] = *(const bf16x8*)(B + (int64_t)(k0 + r) * N + n0 + c8);
      }
    }
  };
  // XOR-swizzled k-granule for the B image: element (n, k) lives at granule
  // (k/8) ^ (n>>3 masked to the granule count).  Without it the TRANS_B
  // scalar transposed writes land all lanes in a handful of banks (same
  // pathology PMC exposed in wgrad: ~12 conflict cycles per LDS op); the
  // b128 fragment reads stay 16-byte aligned.
  auto bswz = [](int n, int k) {
    return n * LK +
           ((((k >> 3) ^ ((n >> 3) & (TBK / 8 - 1))) << 3) | (k & 7));
  };
  auto stage_write = [&](int buf, bf16x8 (&ra)[PA], bf16x8 (&rb)[PB]) {
#pragma unroll
    for (int i = 0; i < PA; ++i) {
      const int p = tid + i * 256;
      const int r = p / ACPR, c8 = (p % ACPR) * 8;
      *(bf16x8*)&lds_a[buf][r * LK + c8] = ra[i];
    }
#pragma unroll
    for (int i = 0; i < PB; ++i) {
      const int p = tid + i * 256;
      if (!TRANS_B) {
        const int r = p / ACPR, c8 = (p % ACPR) * 8;
        *(bf16x8*)&lds_b[buf][bswz(r, c8)] = rb[i];
      } else {
        const int r = p / (TBN / 8), c8 = (p % (TBN / 8)) * 8;  // k-row, n-col
#pragma unroll
        for (int j = 0; j < 8; ++j)
          lds_b[buf][bswz(c8 + j, r)] = rb[i][j];
      }
    }
  };

  int buf = 0;
  {
    bf16x8 ra[PA], rb[PB];
    stage_load(0, ra, rb);
    stage_write(0, ra, rb);
  }
  __syncthreads();

  for (int k0 = 0; k0 < K; k0 += TBK) {
    bf16x8 ra[PA], rb[PB];
    const bool prefetch = k0 + TBK < K;
    if (prefetch) stage_load(k0 + TBK, ra, rb);
#pragma unroll
    for (int kk = 0; kk < TBK; kk += 32) {
#pragma unroll
      for (int am = 0; am < 4; ++am) {
        const bf16x8 a_frag = *(const bf16x8*)
            &lds_a[buf][(wr * 64 + am * 16 + fi) * LK + kk + fk8];
#pragma unroll
        for (int bn = 0; bn < NFRAG; ++bn) {
          const bf16x8 b_frag = *(const bf16x8*)
              &lds_b[buf][bswz(wc * WN + bn * 16 + fi, kk + fk8)];
          acc[am][bn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag, b_frag, acc[am][bn], 0, 0, 0);
        }
      }
    }
    if (prefetch) stage_write(buf ^ 1, ra, rb);
    buf ^= 1;
    __syncthreads();
  }

#pragma unroll
  for (int am = 0; am < 4; ++am) {
#pragma unroll
    for (int bn = 0; bn < NFRAG; ++bn) {
      const int col = n0 + wc * WN + bn * 16 + fi;
      if (col >= N) continue;
      const float bval = bias ? bias[col] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wr * 64 + am * 16 + (lane >> 4) * 4 + r;
        if (row >= M) continue;
        float v = acc[am][bn][r] + bval;
        if (ACT == 1) v = fmaxf(v, 0.0f);
        if (STORE_F32)
          ((float*)C)[(int64_t)row * N + col] = v;
        else
          ((short*)C)[(int64_t)row * N + col] = f2bf(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// wgrad v2: glds-staged tiles + ds_read_b64_tr_b16 hardware transpose reads.
//
// dW[n,k] = sum_m dC[m,n] * A[m,k] reduces over the LEADING dim of both
// operands, so both MFMA fragments need COLUMN reads of row-major tiles —
// the v1 kernel pays for that with 8 scalar transposed ds_writes per loaded
// vector (its measured bottleneck).  Here both tiles stage LINEARLY through
// global_load_lds into 16-column panels and the fragments come from the
// gfx950 transpose-read.  The tr pattern delivers m-elements to lanes in a
// DIFFERENT order than mfma_16x16x32's nominal k-layout — which is
// irrelevant: the reduction dim is order-invariant as long as BOTH operands
// use the same (lane, slot) -> m mapping, and both are read with the same
// instruction from identically-shaped panels.
// Constraints: M % 64 == 0, N % 128 == 0, K % 128 == 0 (launcher falls
// back to v1 otherwise — which keeps the small/odd layers).
using bf16x4t = __attribute__((ext_vector_type(4))) __bf16;

__global__ __launch_bounds__(256) void wgrad_tr_kernel(
    const short* __restrict__ dC, const short* __restrict__ A,
    float* __restrict__ dW, int M, int N, int K, int splitm, int ldw,
    int kmax, int accum) {
  constexpr int MT = 64;  // m-rows per buffer
  // panel layout per image: [8 panels][MT rows][16 cols] bf16 = 16 KB
  __shared__ short lds_c[2][8 * MT * 16];
  __shared__ short lds_a[2][8 * MT * 16];
  const int n_tiles_k = K / 128;
  const int tile_id = blockIdx.x / splitm;
  const int m_part = blockIdx.x % splitm;
  const int n0 = (tile_id / n_tiles_k) * 128;
  const int k0 = (tile_id % n_tiles_k) * 128;
  const int tid = threadIdx.x;
  const int wave = tid / 64, lane = tid % 64;
  const int wr = wave / 2, wc = wave % 2;
  const int fi = lane & 15;

  const int m_chunk = ((M + splitm - 1) / splitm + MT - 1) / MT * MT;
  const int m_begin = m_part * m_chunk;
  const int m_end = min(M, m_begin + m_chunk);

  // glds chunks: wave w stages panels [w*2, w*2+2) of each image (4 x 1 KB
  // per image).  Chunk c of a panel covers m-rows [c*32, c*32+32); lane l
  // holds m = c*32 + l/2, cols (l%2)*8..+8 — 16 contiguous global bytes.
  const int s_m = lane >> 1;
  const int s_n8 = (lane & 1) * 8;
  auto stage = [&](int bufi, int m0g) {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int p = wave * 2 + (c >> 1);
      const int mrow = (c & 1) * 32 + s_m;
      const int64_t m_g = (int64_t)(m0g + mrow);
      __builtin_amdgcn_global_load_lds(
          (const uint32_t*)(dC + m_g * N + n0 + p * 16 + s_n8),
          (uint32_t*)&lds_c[bufi][(p * MT + (c & 1) * 32) * 16], 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const uint32_t*)(A + m_g * K + k0 + p * 16 + s_n8),
          (uint32_t*)&lds_a[bufi][(p * MT + (c & 1) * 32) * 16], 16, 0, 0);
    }
  };
  // transpose-read an 8-m fragment of panel p starting at m-row ms.
  // Measured semantics (tools/trprobe): the 16 lanes of a group concatenate
  // their 8-byte slots into a 4x16 row-major block and each lane receives
  // its column — so with per-lane address (row ms + 8*(lane>>4))*16 +
  // (lane&15)*4 shorts, lane l ends up holding m = ms + 8*(l>>4) + j, the
  // NOMINAL mfma_16x16x32 k-layout (both operands read identically).
  const int t_l4 = (lane & 15) * 4;
  const int t_g8 = (lane >> 4) * 8;
  auto tfrag = [&](const short* img, int p, int ms) {
    const short* b0 = &img[(p * MT + ms + t_g8) * 16 + t_l4];
    bf16x4t lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (__attribute__((address_space(3))) bf16x4t*)(void*)const_cast<short*>(b0));
    bf16x4t hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (__attribute__((address_space(3))) bf16x4t*)(void*)const_cast<short*>(b0 + 4 * 16));
    bf16x8 r;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      r[j] = ((const short*)&lo)[j];
      r[4 + j] = ((const short*)&hi)[j];
    }
    return r;
  };

  f32x4 acc[4][4] = {};
  stage(0, m_begin);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  int buf = 0;
  for (int m0g = m_begin; m0g < m_end; m0g += MT) {
    if (m0g + MT < m_end) stage(buf ^ 1, m0g + MT);
#pragma unroll
    for (int ms = 0; ms < MT; ms += 32) {
      bf16x8 bfr[4];
#pragma unroll
      for (int bk = 0; bk < 4; ++bk)
        bfr[bk] = tfrag(lds_a[buf], wc * 4 + bk, ms);
#pragma unroll
      for (int an = 0; an < 4; ++an) {
        const bf16x8 afr = tfrag(lds_c[buf], wr * 4 + an, ms);
#pragma unroll
        for (int bk = 0; bk < 4; ++bk)
          acc[an][bk] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr, bfr[bk], acc[an][bk], 0, 0, 0);
      }
    }
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    buf ^= 1;
  }

#pragma unroll
  for (int an = 0; an < 4; ++an) {
#pragma unroll
    for (int bk = 0; bk < 4; ++bk) {
      const int n = n0 + wr * 64 + an * 16 + (lane >> 4) * 4;
      const int k = k0 + wc * 64 + bk * 16 + fi;
      if (k >= kmax) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        if (accum || splitm > 1)
          atomicAdd(&dW[(int64_t)(n + r) * ldw + k], acc[an][bk][r]);
        else
          dW[(int64_t)(n + r) * ldw + k] = acc[an][bk][r];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// NT GEMM v2: 2-buffer global_load_lds pipeline (guide §5 "minimum
// 2-phase" + §5.4 rule 21).  The register-staged loop above drains
// vmcnt(0)+lgkmcnt(0) inside every __syncthreads while its stage loads are
// still in flight — PMC on the bench shape showed SQ_WAIT 9:1 over busy and
// ~14% MFMA utilization.  Here the next K-tile's loads go straight to LDS
// via glds DMA issued BEFORE the MFMA block, the barrier is a raw s_barrier
// with explicit counted waits, and the LDS image is lane-linear with the
// conflict-spreading XOR applied to the SOURCE addresses and the fragment
// reads (the same involution on both sides — rule 21).
// Constraints: K % 64 == 0, N % 128 == 0 (launcher falls back otherwise).
template <int ACT, bool STORE_F32, int TBN>
__global__ __launch_bounds__(256) void gemm_nt_glds_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    const float* __restrict__ bias, void* __restrict__ C, int M, int N,
    int K) {
  constexpr int TBK = 64;  // 128 B per image row
  constexpr int WN = TBN / 2;      // per-wave N extent (2x2 wave grid)
  constexpr int NFRAG = WN / 16;
  constexpr int BCH = TBN / 32;    // B glds chunks per wave
  __shared__ short lds_a[2][128 * TBK];
  __shared__ short lds_b[2][TBN * TBK];
  const int n_tiles_n = N / TBN;
  const int m0 = (blockIdx.x / n_tiles_n) * 128;
  const int n0 = (blockIdx.x % n_tiles_n) * TBN;
  const int tid = threadIdx.x;
  const int lane = tid % 64;
  const int wave = tid / 64;
  const int wr = wave / 2, wc = wave % 2;
  const int fi = lane & 15;
  const int g16 = lane >> 4;  // fragment k-granule (16 B units)

  // wave w stages rows [w*32, w*32+32) of each image as 4 glds chunks of
  // 8 rows (64 lanes x 16 B = 1 KB, lane-linear dest).  LDS granule g of
  // row r holds LOGICAL granule g ^ (r & 7): the source address carries the
  // XOR so the read side below is conflict-spread over 8 slots.
  const int s_rl = lane >> 3;       // row within chunk (== row & 7)
  const int s_gr = (lane & 7) ^ s_rl;  // pre-swizzled source granule
  auto stage = [&](int bufi, int k0) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int row = wave * 32 + i * 8 + s_rl;
      const int a_row = min(m0 + row, M - 1);
      __builtin_amdgcn_global_load_lds(
          (const uint32_t*)(A + (int64_t)a_row * K + k0 + s_gr * 8),
          (uint32_t*)&lds_a[bufi][(wave * 32 + i * 8) * TBK], 16, 0, 0);
    }
#pragma unroll
    for (int i = 0; i < BCH; ++i) {
      const int brow = wave * (TBN / 4) + i * 8 + s_rl;
      __builtin_amdgcn_global_load_lds(
          (const uint32_t*)(B + (int64_t)(n0 + brow) * K + k0 + s_gr * 8),
          (uint32_t*)&lds_b[bufi][(wave * (TBN / 4) + i * 8) * TBK], 16, 0, 0);
    }
  };
  auto frag = [&](const short* base, int row, int gr) {
    return *(const bf16x8*)&base[row * TBK + ((gr ^ (row & 7)) << 3)];
  };

  f32x4 acc[4][NFRAG] = {};
  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  int buf = 0;
  for (int k0 = 0; k0 < K; k0 += TBK) {
    if (k0 + TBK < K) stage(buf ^ 1, k0 + TBK);  // DMA hides under MFMA
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 bfr[NFRAG];
#pragma unroll
      for (int bn = 0; bn < NFRAG; ++bn)
        bfr[bn] = frag(lds_b[buf], wc * WN + bn * 16 + fi, kk * 4 + g16);
#pragma unroll
      for (int am = 0; am < 4; ++am) {
        const bf16x8 afr = frag(lds_a[buf], wr * 64 + am * 16 + fi,
                                kk * 4 + g16);
#pragma unroll
        for (int bn = 0; bn < NFRAG; ++bn)
          acc[am][bn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr, bfr[bn], acc[am][bn], 0, 0, 0);
      }
    }
    // next tile's glds must have LANDED (vmcnt) and this tile's reads be
    // done (lgkm) before any wave re-stages over them
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    buf ^= 1;
  }

#pragma unroll
  for (int am = 0; am < 4; ++am) {
#pragma unroll
    for (int bn = 0; bn < NFRAG; ++bn) {
      const int col = n0 + wc * WN + bn * 16 + fi;
      const float bval = bias ? bias[col] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wr * 64 + am * 16 + (lane >> 4) * 4 + r;
        if (row >= M) continue;
        float v = acc[am][bn][r] + bval;
        if (ACT == 1) v = fmaxf(v, 0.0f);
        if (STORE_F32)
          ((float*)C)[(int64_t)row * N + col] = v;
        else
          ((short*)C)[(int64_t)row * N + col] = f2bf(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// NT GEMM v3: 3-buffer glds ring with COUNTED vmcnt — one K-tile stays in
// flight ACROSS each barrier (guide §2 "3-buf span": +83% over serial vs
// +40% for the 2-buf drain; the v2 kernel above waits vmcnt(0) per tile).
// 96 KB LDS -> 1 block/CU, the regime where glds pipelining pays.
// 8 glds per wave per tile => s_waitcnt vmcnt(8) keeps exactly the newest
// tile's loads outstanding.
template <int ACT, bool STORE_F32>
__global__ __launch_bounds__(256) void gemm_nt_glds3_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    const float* __restrict__ bias, void* __restrict__ C, int M, int N,
    int K) {
  constexpr int TBK = 64;
  __shared__ short lds_a[3][128 * TBK];
  __shared__ short lds_b[3][128 * TBK];
  const int n_tiles_n = N / 128;
  const int m0 = (blockIdx.x / n_tiles_n) * 128;
  const int n0 = (blockIdx.x % n_tiles_n) * 128;
  const int tid = threadIdx.x;
  const int lane = tid % 64;
  const int wave = tid / 64;
  const int wr = wave / 2, wc = wave % 2;
  const int fi = lane & 15;
  const int g16 = lane >> 4;
  const int s_rl = lane >> 3;
  const int s_gr = (lane & 7) ^ s_rl;
  auto stage = [&](int bufi, int k0) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int row = wave * 32 + i * 8 + s_rl;
      const int a_row = min(m0 + row, M - 1);
      __builtin_amdgcn_global_load_lds(
          (const uint32_t*)(A + (int64_t)a_row * K + k0 + s_gr * 8),
          (uint32_t*)&lds_a[bufi][(wave * 32 + i * 8) * TBK], 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const uint32_t*)(B + (int64_t)(n0 + row) * K + k0 + s_gr * 8),
          (uint32_t*)&lds_b[bufi][(wave * 32 + i * 8) * TBK], 16, 0, 0);
    }
  };
  auto frag = [&](const short* base, int row, int gr) {
    return *(const bf16x8*)&base[row * TBK + ((gr ^ (row & 7)) << 3)];
  };

  f32x4 acc[4][4] = {};
  const int nt = K / TBK;
  stage(0, 0);
  if (nt > 1) {
    stage(1, TBK);
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");  // tile 0 landed
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();
  for (int t = 0; t < nt; ++t) {
    if (t + 2 < nt) stage((t + 2) % 3, (t + 2) * TBK);
    const short* la = lds_a[t % 3];
    const short* lb = lds_b[t % 3];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 bfr[4];
#pragma unroll
      for (int bn = 0; bn < 4; ++bn)
        bfr[bn] = frag(lb, wc * 64 + bn * 16 + fi, kk * 4 + g16);
#pragma unroll
      for (int am = 0; am < 4; ++am) {
        const bf16x8 afr = frag(la, wr * 64 + am * 16 + fi, kk * 4 + g16);
#pragma unroll
        for (int bn = 0; bn < 4; ++bn)
          acc[am][bn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr, bfr[bn], acc[am][bn], 0, 0, 0);
      }
    }
    // keep the NEWEST tile's 8 glds in flight across the barrier; the next
    // iteration's buffer (t+1) is guaranteed landed.  Near the tail nothing
    // newer is in flight, so drain fully before the final tiles.
    if (t + 2 < nt)
      asm volatile("s_waitcnt vmcnt(8) lgkmcnt(0)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

#pragma unroll
  for (int am = 0; am < 4; ++am) {
#pragma unroll
    for (int bn = 0; bn < 4; ++bn) {
      const int col = n0 + wc * 64 + bn * 16 + fi;
      const float bval = bias ? bias[col] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wr * 64 + am * 16 + (lane >> 4) * 4 + r;
        if (row >= M) continue;
        float v = acc[am][bn][r] + bval;
        if (ACT == 1) v = fmaxf(v, 0.0f);
        if (STORE_F32)
          ((float*)C)[(int64_t)row * N + col] = v;
        else
          ((short*)C)[(int64_t)row * N + col] = f2bf(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// wgrad: dW[N,K] += sum_m dC[m][n] * A[m][k]   (dC [M,N], A [M,K] bf16;
// dW f32, torch Linear weight layout).  MFMA with the reduction on the
// memory-row axis: both tiles are staged TRANSPOSED ([n][m] / [k][m],
// scalar writes in the staging pass) so fragment reads are contiguous.
// 64x64 (n,k) tiles, 4 waves as 2x2 of 32x32; M split across blocks with
// atomicAdd (small-K layers underfill the chip otherwise).
// ---------------------------------------------------------------------------
constexpr int WTM = 64;       // m chunk per stage (2 MFMA k-steps per barrier)
constexpr int WLD = WTM + PAD;

__global__ __launch_bounds__(256) void wgrad_kernel(
    const short* __restrict__ dC, const short* __restrict__ A,
    float* __restrict__ dW, int M, int N, int K, int splitm, int ldw,
    int kmax, int accum) {
  __shared__ short lds_dct[2][64 * WLD];  // [n][m]
  __shared__ short lds_at[2][64 * WLD];   // [k][m]
  // staging pieces: [WTM rows x 64 cols] / (256 threads x 8 shorts) = 2 each

  const int n_tiles_k = (K + 63) / 64;
  const int tile_id = blockIdx.x / splitm;
  const int m_part = blockIdx.x % splitm;
  const int n0 = (tile_id / n_tiles_k) * 64;
  const int k0 = (tile_id % n_tiles_k) * 64;

  const int tid = threadIdx.x;
  const int wave = tid / 64, lane = tid % 64;
  const int wr = wave / 2, wc = wave % 2;
  const int fi = lane & 15;
  const int fk8 = (lane >> 4) * 8;

  const int m_chunk = ((M + splitm - 1) / splitm + WTM - 1) / WTM * WTM;
  const int m_begin = m_part * m_chunk;
  const int m_end = min(M, m_begin + m_chunk);

  f32x4 acc[2][2] = {};

  // staging: thread loads 8 contiguous cols of one m-row, writes transposed
  // (T14 split: loads issued before the MFMA block, writes after)
  const int s_m = tid / 8, s_c8 = (tid % 8) * 8;  // 2x 32 m-rows x 64 cols

  auto wload = [&](int m0, bf16x8 (&v)[2], bf16x8 (&w)[2]) {
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int m = m0 + s_m + h * 32;
      v[h] = bf16x8{};
      if (m < M && n0 + s_c8 < N)
        v[h] = *(const bf16x8*)(dC + (int64_t)m * N + n0 + s_c8);
      w[h] = bf16x8{};
      if (m < M && k0 + s_c8 < K)
        w[h] = *(const bf16x8*)(A + (int64_t)m * K + k0 + s_c8);
    }
  };
  // XOR-swizzled m-granule: element (n, m) lives at granule (m/8)^((n>>3)&7)
  // inside row n.  The plain transposed scalar writes put all 64 lanes in 4
  // banks (PMC: ~12 conflict cycles per LDS instruction); the swizzle
  // spreads simultaneous writes across the full bank set (~2-way) while the
  // b128 fragment reads keep their 16-byte alignment and 2-way pattern.
  auto wswz = [](int n, int m) {
    return n * WLD + ((((m >> 3) ^ ((n >> 3) & 7)) << 3) | (m & 7));
  };
  auto wwrite = [&](int buf, bf16x8 (&v)[2], bf16x8 (&w)[2]) {
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int m = s_m + h * 32;
#pragma unroll
      for (int j = 0; j < 8; ++j) lds_dct[buf][wswz(s_c8 + j, m)] = v[h][j];
#pragma unroll
      for (int j = 0; j < 8; ++j) lds_at[buf][wswz(s_c8 + j, m)] = w[h][j];
    }
  };

  int buf = 0;
  {
    bf16x8 v[2], w[2];
    wload(m_begin, v, w);
    wwrite(0, v, w);
  }
  __syncthreads();

  for (int m0 = m_begin; m0 < m_end; m0 += WTM) {
    bf16x8 v[2], w[2];
    const bool prefetch = m0 + WTM < m_end;
    if (prefetch) wload(m0 + WTM, v, w);
#pragma unroll
    for (int ms = 0; ms < WTM; ms += 32) {
#pragma unroll
      for (int an = 0; an < 2; ++an) {
        const bf16x8 a_frag = *(const bf16x8*)
            &lds_dct[buf][wswz(wr * 32 + an * 16 + fi, ms + fk8)];
#pragma unroll
        for (int bk = 0; bk < 2; ++bk) {
          const bf16x8 b_frag = *(const bf16x8*)
              &lds_at[buf][wswz(wc * 32 + bk * 16 + fi, ms + fk8)];
          acc[an][bk] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag, b_frag, acc[an][bk], 0, 0, 0);
        }
      }
    }
    if (prefetch) wwrite(buf ^ 1, v, w);
    buf ^= 1;
    __syncthreads();
  }

#pragma unroll
  for (int an = 0; an < 2; ++an) {
#pragma unroll
    for (int bk = 0; bk < 2; ++bk) {
      const int k = k0 + wc * 32 + bk * 16 + fi;
      if (k >= kmax) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int n = n0 + wr * 32 + an * 16 + (lane >> 4) * 4 + r;
        if (n >= N) continue;
        if (accum || splitm > 1)
          atomicAdd(&dW[(int64_t)n * ldw + k], acc[an][bk][r]);
        else
          dW[(int64_t)n * ldw + k] = acc[an][bk][r];
      }
    }
  }
}

// ------------------------------------------------------- fused BCE-with-logits
// loss = mean_i [ max(z,0) - z*y + log1p(exp(-|z|)) ]; one pass computes the
// per-block partial sums (atomic scalar add) AND caches sigmoid(z) for the
// backward, which is a single elementwise kernel: dz = g * (sigmoid(z)-y)/B.
// Replaces torch's ~10-kernel BCEWithLogits fwd+bwd chain in the captured
// step (each small kernel costs ~5 us of launch-bound GPU time in a graph).
__global__ void bce_fwd_kernel(const float* __restrict__ z,
                               const float* __restrict__ y,
                               float* __restrict__ sig,
                               float* __restrict__ loss_sum, int64_t n) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  float l = 0.0f;
  if (i < n) {
    const float zi = z[i], yi = y[i];
    sig[i] = 1.0f / (1.0f + __expf(-zi));
    l = fmaxf(zi, 0.0f) - zi * yi + __logf(1.0f + __expf(-fabsf(zi)));
  }
  // block reduce via LDS then one atomic per block
  __shared__ float part[256];
  part[threadIdx.x] = l;
  __syncthreads();
  for (int s = 128; s > 0; s >>= 1) {
    if (threadIdx.x < s) part[threadIdx.x] += part[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(loss_sum, part[0]);
}

__global__ void bce_bwd_kernel(const float* __restrict__ sig,
                               const float* __restrict__ y,
                               const float* __restrict__ gscale,
                               float* __restrict__ dz, int64_t n) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  dz[i] = gscale[0] * (sig[i] - y[i]) / (float)n;
}

// ---------------------------------------------------------------------------
__global__ void relu_bwd_kernel(const short* __restrict__ g,
                                const short* __restrict__ out,
                                short* __restrict__ g_eff, int64_t n) {
  const int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (i >= n) return;
  bf16x8 gv = *(const bf16x8*)(g + i);
  bf16x8 ov = *(const bf16x8*)(out + i);
  bf16x8 r;
#pragma unroll
  for (int j = 0; j < 8; ++j) r[j] = bf2f(ov[j]) > 0.0f ? gv[j] : (short)0;
  *(bf16x8*)(g_eff + i) = r;
}

// relu backward FUSED with the bias column-sum: the masked gradient is the
// exact tensor bias_grad would re-read (16 MB round-trip + a launch per
// layer saved).  Same column-invariance + LDS-reduce + small atomic fan-in
// structure as bias_grad_vec_kernel; power-of-two N only.
__global__ __launch_bounds__(256) void relu_bwd_bias_kernel(
    const short* __restrict__ g, const short* __restrict__ out,
    short* __restrict__ g_eff, float* __restrict__ db, int64_t total8,
    int N) {
  extern __shared__ float lds_col[];  // [N]
  const int n8 = N / 8;
  for (int i = threadIdx.x; i < N; i += blockDim.x) lds_col[i] = 0.0f;
  __syncthreads();
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int c8 = (int)((idx % n8) * 8);  // invariant: stride % n8 == 0
  float acc[8] = {};
  for (; idx < total8; idx += stride) {
    const bf16x8 gv = *(const bf16x8*)(g + idx * 8);
    const bf16x8 ov = *(const bf16x8*)(out + idx * 8);
    bf16x8 r;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      r[j] = bf2f(ov[j]) > 0.0f ? gv[j] : (short)0;
      acc[j] += bf2f(r[j]);
    }
    *(bf16x8*)(g_eff + idx * 8) = r;
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) atomicAdd(&lds_col[c8 + j], acc[j]);
  __syncthreads();
  for (int i = threadIdx.x; i < N; i += blockDim.x)
    atomicAdd(&db[i], lds_col[i]);
}

// db[n] = sum_m dC[m][n]: each block reduces a chunk of rows over ALL cols
// (threads stride the row -> fully coalesced); one atomicAdd per (block,col).
__global__ void bias_grad_kernel(const short* __restrict__ dC,
                                 float* __restrict__ db, int M, int N,
                                 int rows_per_block) {
  const int m_begin = blockIdx.x * rows_per_block;
  const int m_end = min(M, m_begin + rows_per_block);
  for (int col = threadIdx.x; col < N; col += blockDim.x) {
    float acc = 0.0f;
    for (int m = m_begin; m < m_end; ++m) acc += bf2f(dC[(int64_t)m * N + col]);
    atomicAdd(&db[col], acc);
  }
}

// Vector column-sum for power-of-two N (the DLRM layer widths): the matrix
// is consumed as flat bf16x8 vectors with FULLY COALESCED 16-byte lane
// loads; because the grid-stride (grid*256*8) is a multiple of N, every
// thread's vector always covers the SAME 8 columns, so the whole pass
// accumulates in 8 registers.  A block then reduces its 2048 lane-octets
// into N partials through LDS and issues one atomicAdd per column.  The
// scalar kernel above (2 B per lane per load) measured 24.5 us for an
// 8 MB matrix — ~20x off the bandwidth bound this version targets.
__global__ __launch_bounds__(256) void bias_grad_vec_kernel(
    const short* __restrict__ dC, float* __restrict__ db, int64_t total8,
    int N) {
  extern __shared__ float lds_col[];  // [N]
  const int n8 = N / 8;
  for (int i = threadIdx.x; i < N; i += blockDim.x) lds_col[i] = 0.0f;
  __syncthreads();
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int c8 = (int)((idx % n8) * 8);  // invariant: stride % n8 == 0
  float acc[8] = {};
  // 4 independent vector loads in flight per iteration: with a modest grid
  // (atomic fan-in per column is gridDim serialized in L2) the kernel needs
  // unrolled MLP to reach HBM bandwidth
  constexpr int U = 4;
  for (; idx + 3 * stride < total8; idx += U * stride) {
    bf16x8 v[U];
#pragma unroll
    for (int t = 0; t < U; ++t)
      v[t] = *(const bf16x8*)(dC + (idx + t * stride) * 8);
#pragma unroll
    for (int t = 0; t < U; ++t)
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += bf2f(v[t][j]);
  }
  for (; idx < total8; idx += stride) {
    const bf16x8 v = *(const bf16x8*)(dC + idx * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += bf2f(v[j]);
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) atomicAdd(&lds_col[c8 + j], acc[j]);
  __syncthreads();
  for (int i = threadIdx.x; i < N; i += blockDim.x)
    atomicAdd(&db[i], lds_col[i]);
}

}  // namespace

// ============================================================ torch bindings

static hipStream_t dcur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

// C = act(A @ op(B) + bias); A [M,K] bf16.
// trans_b == 0: B is [N, K] (torch weight layout), C = A @ B^T.
// trans_b == 1: B is [K, N] row-major, C = A @ B (the dgrad case: pass the
//               [N_layer, K_layer] weight directly as the [K,N] operand).
torch::Tensor gemm_nt_bias_act(torch::Tensor A, torch::Tensor B,
                               torch::Tensor bias, int64_t act,
                               int64_t out_f32, int64_t trans_b) {
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16 &&
              B.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
  const int M = (int)A.size(0), K = (int)A.size(1);
  const int N = (int)(trans_b ? B.size(1) : B.size(0));
  TORCH_CHECK((trans_b ? B.size(0) : B.size(1)) == K, "gemm_nt shape mismatch");
  TORCH_CHECK(K % BK == 0, "K must be a multiple of 32 (pad)");
  TORCH_CHECK(N % 8 == 0, "N must be a multiple of 8");
  auto C = torch::empty(
      {M, N},
      torch::TensorOptions()
          .dtype(out_f32 ? torch::kFloat32 : torch::kBFloat16)
          .device(A.device()));
  const int m_tiles = (M + BM - 1) / BM;
  // v2 glds pipeline when shapes allow (PA_GEMM_V2=0 reverts)
  static const bool v2_on = [] {
    const char* e = getenv("PA_GEMM_V2");
    return !e || atoi(e) != 0;
  }();
  // shape cutover (measured sweep, tools/gemm_v2_sweep.py): v2 wins when the
  // 128x128 grid fills the chip (>=400 blocks) and K amortizes the pipeline
  // prologue; below that the v1 TBN=64 configs keep more CUs busy
  const bool v2_shape = v2_on && !trans_b && K % 64 == 0 && M >= 128 &&
                        K >= 256;
  const bool v2_128 = v2_shape && N % 128 == 0 &&
                      (int64_t)m_tiles * (N / 128) >= 400;
  // TBN=64 variant: mid-width layers whose 128-wide grid underfills
  const bool v2_64 = v2_shape && !v2_128 && N % 64 == 0 &&
                     (int64_t)m_tiles * (N / 64) >= 400;
  static const bool v3_on = [] {
    const char* e = getenv("PA_GEMM_V3");
    return e && atoi(e) != 0;
  }();
  if (v3_on && v2_128) {
    const int grid = m_tiles * (N / 128);
    const float* bias_p = bias.numel() ? bias.data_ptr<float>() : nullptr;
#define PA_GEMM3(ACTV, F32V)                                                  \
  hipLaunchKernelGGL((gemm_nt_glds3_kernel<ACTV, F32V>), dim3(grid),          \
                     dim3(256), 0, dcur_stream(),                             \
                     (const bf16*)A.data_ptr(), (const bf16*)B.data_ptr(),    \
                     bias_p, C.data_ptr(), M, N, K)
    if (act == 1) { if (out_f32) PA_GEMM3(1, true); else PA_GEMM3(1, false); }
    else          { if (out_f32) PA_GEMM3(0, true); else PA_GEMM3(0, false); }
#undef PA_GEMM3
    return C;
  }
  if (v2_128 || v2_64) {
    const int tbn = v2_128 ? 128 : 64;
    const int grid = m_tiles * (N / tbn);
    const float* bias_p = bias.numel() ? bias.data_ptr<float>() : nullptr;
#define PA_GEMM2(ACTV, F32V, TBNV)                                            \
  hipLaunchKernelGGL((gemm_nt_glds_kernel<ACTV, F32V, TBNV>), dim3(grid),     \
                     dim3(256), 0, dcur_stream(),                             \
                     (const bf16*)A.data_ptr(), (const bf16*)B.data_ptr(),    \
                     bias_p, C.data_ptr(), M, N, K)
#define PA_GEMM2_T(ACTV, F32V)                                                \
  do {                                                                        \
    if (v2_128) PA_GEMM2(ACTV, F32V, 128);                                    \
    else PA_GEMM2(ACTV, F32V, 64);                                            \
  } while (0)
    if (act == 1) { if (out_f32) PA_GEMM2_T(1, true); else PA_GEMM2_T(1, false); }
    else          { if (out_f32) PA_GEMM2_T(0, true); else PA_GEMM2_T(0, false); }
#undef PA_GEMM2_T
#undef PA_GEMM2
    return C;
  }
  // tile config: BK=64 halves barriers (needs K%64); BN=64 doubles the grid
  // (skinny layers underfill 256 CUs at BN=128); (128,64) exceeds 64KB LDS.
  const bool k64 = K % 64 == 0 && N % 64 == 0;
  const bool narrow =
      ((int64_t)m_tiles * ((N + 127) / 128) < 192 && N % 64 == 0) || k64;
  const int tbn = narrow ? 64 : 128;
  const int grid = m_tiles * ((N + tbn - 1) / tbn);
  const float* bias_ptr = bias.numel() ? bias.data_ptr<float>() : nullptr;
  hipStream_t st = dcur_stream();
#define PA_GEMM(ACTV, F32V, TBV, TBNV, TBKV)                                  \
  hipLaunchKernelGGL((gemm_nt_kernel<ACTV, F32V, TBV, TBNV, TBKV>),           \
                     dim3(grid), dim3(256), 0, st, (const bf16*)A.data_ptr(), \
                     (const bf16*)B.data_ptr(), bias_ptr, C.data_ptr(), M, N, \
                     K)
#define PA_GEMM_CFG(ACTV, F32V, TBV)                                          \
  do {                                                                        \
    if (k64) PA_GEMM(ACTV, F32V, TBV, 64, 64);                                \
    else if (narrow) PA_GEMM(ACTV, F32V, TBV, 64, 32);                        \
    else PA_GEMM(ACTV, F32V, TBV, 128, 32);                                   \
  } while (0)
#define PA_GEMM_T(ACTV, F32V)                                                 \
  do {                                                                        \
    if (trans_b) PA_GEMM_CFG(ACTV, F32V, true);                               \
    else PA_GEMM_CFG(ACTV, F32V, false);                                      \
  } while (0)
  if (act == 1) { if (out_f32) PA_GEMM_T(1, true); else PA_GEMM_T(1, false); }
  else          { if (out_f32) PA_GEMM_T(0, true); else PA_GEMM_T(0, false); }
#undef PA_GEMM_T
#undef PA_GEMM_CFG
#undef PA_GEMM
  return C;
}

// -> g_eff; db accumulated into the given f32 buffer (pre-zeroed slot or
// torch.zeros).  Falls back to the unfused pair for non-pow2 N.
torch::Tensor relu_bwd_bias(torch::Tensor g, torch::Tensor out,
                            torch::Tensor db) {
  const int64_t M = g.size(0), N = g.size(1);
  TORCH_CHECK(db.scalar_type() == torch::kFloat32 && db.numel() == N);
  const bool pow2 = (N & (N - 1)) == 0 && N >= 8 && N <= 8192;
  TORCH_CHECK(pow2, "relu_bwd_bias: power-of-two N only");
  auto g_eff = torch::empty_like(g);
  const int64_t total8 = M * N / 8;
  const int n8 = (int)(N / 8);
  int blocks = (int)std::min<int64_t>(256, (total8 + 255) / 256);
  if (n8 > 256) blocks = ((blocks + n8 / 256 - 1) / (n8 / 256)) * (n8 / 256);
  hipLaunchKernelGGL(relu_bwd_bias_kernel, dim3(blocks), dim3(256),
                     (int)N * sizeof(float), dcur_stream(),
                     (const short*)g.data_ptr(), (const short*)out.data_ptr(),
                     (short*)g_eff.data_ptr(), db.data_ptr<float>(), total8,
                     (int)N);
  return g_eff;
}

torch::Tensor relu_bwd(torch::Tensor g, torch::Tensor out) {
  auto g_eff = torch::empty_like(g);
  const int64_t n = g.numel();
  TORCH_CHECK(n % 8 == 0);
  const int64_t blocks = (n / 8 + 255) / 256;
  hipLaunchKernelGGL(relu_bwd_kernel,
                     dim3((unsigned)std::min<int64_t>(blocks, 1 << 30)),
                     dim3(256), 0, dcur_stream(), (const short*)g.data_ptr(),
                     (const short*)out.data_ptr(), (short*)g_eff.data_ptr(), n);
  return g_eff;
}

static void bias_grad_launch(torch::Tensor dC, torch::Tensor db) {
  const int M = (int)dC.size(0), N = (int)dC.size(1);
  // power-of-two N (every DLRM/DCN layer width): coalesced vector kernel.
  // grid*256 must be a multiple of N/8 so each thread's column octet is
  // loop-invariant; N/8 is a power of two <= 256's multiples, so any grid
  // multiple of max(1, N/2048) works — use a 512-block grid (grid%*)
  const bool pow2 = (N & (N - 1)) == 0 && N >= 8 && N <= 8192;
  if (pow2) {
    const int64_t total8 = (int64_t)M * N / 8;
    // modest grid: every block ends with one global atomicAdd per column,
    // serialized per address in L2 — 512 blocks measured 19 us for an 8 MB
    // matrix; 160 blocks + 4-deep unrolled loads hits the bandwidth bound
    static const int kBlocks = [] {
      const char* e = getenv("PA_BIAS_BLOCKS");
      return e ? atoi(e) : 96;  // swept 64..1024: flat minimum 64-160
    }();
    int blocks = (int)std::min<int64_t>(kBlocks, (total8 + 255) / 256);
    const int n8 = N / 8;
    // round blocks up so (blocks*256) % n8 == 0 (n8 is a power of two)
    if (n8 > 256) blocks = ((blocks + n8 / 256 - 1) / (n8 / 256)) * (n8 / 256);
    hipLaunchKernelGGL(bias_grad_vec_kernel, dim3(blocks), dim3(256),
                       N * sizeof(float), dcur_stream(),
                       (const short*)dC.data_ptr(), db.data_ptr<float>(),
                       total8, N);
    return;
  }
  const int rows_per_block = std::max(8, (M + 511) / 512);
  const int blocks = (M + rows_per_block - 1) / rows_per_block;
  hipLaunchKernelGGL(bias_grad_kernel, dim3(blocks), dim3(256), 0,
                     dcur_stream(), (const short*)dC.data_ptr(),
                     db.data_ptr<float>(), M, N, rows_per_block);
}

torch::Tensor bias_grad(torch::Tensor dC) {
  auto db = torch::zeros(
      {dC.size(1)},
      torch::TensorOptions().dtype(torch::kFloat32).device(dC.device()));
  bias_grad_launch(dC, db);
  return db;
}

// Side-band: atomically ACCUMULATE the column sums into the caller's
// pre-zeroed f32 gradient slot (both kernels end in atomicAdd already).
void bias_grad_into(torch::Tensor dC, torch::Tensor out) {
  TORCH_CHECK(out.scalar_type() == torch::kFloat32 && out.is_contiguous());
  TORCH_CHECK(out.numel() == dC.size(1));
  bias_grad_launch(dC, out);
}

// dW [N,K] f32 = dC^T @ A
static void wgrad_launch(torch::Tensor dC, torch::Tensor A, torch::Tensor out,
                         int ldw, int kmax, int accum) {
  const int M = (int)dC.size(0), N = (int)dC.size(1), K = (int)A.size(1);
  static const bool tr_on = [] {
    const char* e = getenv("PA_WGRAD_TR");
    return !e || atoi(e) != 0;
  }();
  // cutover (tools/wgrad_tr_probe.py): the 128x128-tile tr kernel wins only
  // when its own grid fills the chip (1024x1024: 48.7 vs 66.0 us); below
  // that the v1 64x64 tiles + split-M keep more CUs busy
  if (tr_on && M % 64 == 0 && N % 128 == 0 && K % 128 == 0 &&
      (N / 128) * (K / 128) >= 64) {
    const int tiles2 = (N / 128) * (K / 128);
    int splitm = 1;
    while (tiles2 * splitm < 512 && splitm < 64 && (M / (splitm * 2)) >= 64)
      splitm *= 2;
    if (splitm > 1 && !accum) out.zero_();
    hipLaunchKernelGGL(wgrad_tr_kernel, dim3(tiles2 * splitm), dim3(256), 0,
                       dcur_stream(), (const short*)dC.data_ptr(),
                       (const short*)A.data_ptr(), out.data_ptr<float>(), M, N,
                       K, splitm, ldw, kmax, accum);
    return;
  }
  const int tiles = ((N + 63) / 64) * ((K + 63) / 64);
  int splitm = 1;
  while (tiles * splitm < 512 && splitm < 64 && (M / (splitm * 2)) >= 32)
    splitm *= 2;
  if (splitm > 1 && !accum) out.zero_();
  hipLaunchKernelGGL(wgrad_kernel, dim3(tiles * splitm), dim3(256), 0,
                     dcur_stream(), (const short*)dC.data_ptr(),
                     (const short*)A.data_ptr(), out.data_ptr<float>(), M, N,
                     K, splitm, ldw, kmax, accum);
}

torch::Tensor wgrad(torch::Tensor dC, torch::Tensor A) {
  const int N = (int)dC.size(1), K = (int)A.size(1);
  auto dW = torch::empty(
      {N, K}, torch::TensorOptions().dtype(torch::kFloat32).device(A.device()));
  wgrad_launch(dC, A, dW, K, K, 0);
  return dW;
}

// Side-band accumulation: dW atomically ADDS into `out` (f32 [N, kmax],
// row stride = out's width), the caller's pre-zeroed flat gradient slot —
// no dW materialization, no cast, no pad-column slice, no AccumulateGrad
// add.  `out` may be narrower than A (kmax < K drops the zero pad columns).
void wgrad_into(torch::Tensor dC, torch::Tensor A, torch::Tensor out) {
  TORCH_CHECK(out.scalar_type() == torch::kFloat32 && out.is_contiguous());
  TORCH_CHECK(out.size(0) == dC.size(1) && out.size(1) <= A.size(1));
  wgrad_launch(dC, A, out, (int)out.size(1), (int)out.size(1), 1);
}

std::vector<torch::Tensor> bce_fwd(torch::Tensor z, torch::Tensor y) {
  TORCH_CHECK(z.scalar_type() == torch::kFloat32 && y.scalar_type() == torch::kFloat32);
  const int64_t n = z.numel();
  auto sig = torch::empty_like(z);
  auto loss = torch::zeros({1}, z.options());
  hipLaunchKernelGGL(bce_fwd_kernel, dim3((unsigned)((n + 255) / 256)),
                     dim3(256), 0, dcur_stream(), z.data_ptr<float>(),
                     y.data_ptr<float>(), sig.data_ptr<float>(),
                     loss.data_ptr<float>(), n);
  loss.div_((double)n);
  // NOT loss.view({}): the empty brace list resolves to the
  // view(ScalarType) overload (uint8 reinterpret), not a 0-dim view
  return {loss.squeeze(0), sig};
}

torch::Tensor bce_bwd(torch::Tensor sig, torch::Tensor y, torch::Tensor g) {
  const int64_t n = sig.numel();
  auto dz = torch::empty_like(sig);
  hipLaunchKernelGGL(bce_bwd_kernel, dim3((unsigned)((n + 255) / 256)),
                     dim3(256), 0, dcur_stream(), sig.data_ptr<float>(),
                     y.data_ptr<float>(), g.data_ptr<float>(),
                     dz.data_ptr<float>(), n);
  return dz;
}

void init_dense(pybind11::module_& m) {
  m.def("gemm_nt_bias_act", &gemm_nt_bias_act,
        "bf16 MFMA GEMM (A @ B^T) with fused bias+activation");
  m.def("relu_bwd", &relu_bwd, "g * (out > 0)");
  m.def("relu_bwd_bias", &relu_bwd_bias,
        "relu backward fused with the bias column-sum accumulation");
  m.def("bce_fwd", &bce_fwd, "fused BCE-with-logits forward (loss + sigmoid cache)");
  m.def("bce_bwd", &bce_bwd, "fused BCE-with-logits backward");
  m.def("bias_grad", &bias_grad, "column-sum bias gradient");
  m.def("wgrad", &wgrad, "dW = dC^T @ A (f32 out)");
  m.def("wgrad_into", &wgrad_into,
        "dW accumulated straight into a flat f32 grad slot");
  m.def("bias_grad_into", &bias_grad_into,
        "bias grad accumulated straight into a flat f32 grad slot");
}
