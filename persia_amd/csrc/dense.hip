// Fused dense-MLP kernels for gfx950 (CDNA4): bf16 MFMA GEMM with fused
// bias + ReLU epilogues, plus the backward (dgrad / wgrad / bias-grad)
// kernels.  Replaces the reference's dense path (plain torch Linear over
// hipBLASLt + an elementwise cascade) with MFMA/LDS-tiled kernels
// (BASELINE north star: "dense-side fused MLP ... hand-written CDNA4 HIP
// kernels with MFMA/LDS tiling").
//
// Structure follows the CDNA HIP guide §5 canonical GEMM anatomy:
// 128x128 output tile, BK=32 K-step, 4 waves x (64x64 per-wave tile of
// 4x4 mfma_f32_16x16x32_bf16 fragments), double-buffered LDS staged with
// global_load_lds (16B), XOR-swizzled LDS images on the source address.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

using bf16 = __hip_bfloat16;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) short;

namespace {

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int WARPS = 4;  // 2x2 wave grid, 64x64 per wave

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
// fwd: C[M,N] = act(A[M,K] @ B[K,N] + bias[N]); A,B,C bf16 row-major.
// ACT: 0 = none, 1 = relu.
// Each block: 128x128 C-tile; LDS: A-tile [BM][BK], B-tile [BK][BN].
// ---------------------------------------------------------------------------
template <int ACT, bool STORE_F32>
__global__ __launch_bounds__(256) void gemm_bias_act_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    const float* __restrict__ bias, void* __restrict__ C, int M, int N, int K,
    int ldc) {
  __shared__ short lds_a[2][BM * BK];
  __shared__ short lds_b[2][BK * BN];

  const int n_tiles_n = (N + BN - 1) / BN;
  const int tile_m = blockIdx.x / n_tiles_n;
  const int tile_n = blockIdx.x % n_tiles_n;
  const int m0 = tile_m * BM, n0 = tile_n * BN;

  const int tid = threadIdx.x;
  const int wave = tid / 64, lane = tid % 64;
  const int wr = wave / 2, wc = wave % 2;  // wave tile (64x64) position

  // fragment indexing for mfma_f32_16x16x32_bf16:
  //   A frag: lane holds A[i = lane&15][k = (lane>>4)*8 + j], j=0..7
  //   B frag: lane holds B[k = (lane>>4)*8 + j][n = lane&15]
  //   C frag: lane holds C[row = (lane>>4)*4 + r][col = lane&15], r=0..3
  const int fi = lane & 15;        // row/col within fragment
  const int fk8 = (lane >> 4) * 8; // k-offset of this lane's 8 elements

  f32x4 acc[4][4] = {};

  // staging: 256 threads load BM*BK (=4096) A shorts -> 16 shorts each -> but
  // glds moves 16B (8 shorts) per lane; 2 rounds for A, 2 for B.
  // A-tile global: A[m0+r][k0+c]; LDS linear [r][c] (row-major, BK=32 shorts
  // = 64B rows). B-tile: B[k0+r][n0+c], LDS [r][c] (BN=128 shorts = 256B).
  const int a_r = tid / 4, a_c8 = (tid % 4) * 8;     // 64 rows per round
  const int b_r = tid / 16, b_c8 = (tid % 16) * 8;   // 16 rows per round

  int buf = 0;
  // prologue: stage k0 = 0
  {
    const int k0 = 0;
    for (int half = 0; half < 2; ++half) {
      const int r = a_r + half * 64;
      const short* src = (const short*)(A + (int64_t)(m0 + r) * K + k0 + a_c8);
      short* dst = &lds_a[buf][r * BK + a_c8];
      if (m0 + r < M) *(bf16x8*)dst = *(const bf16x8*)src;
      else *(bf16x8*)dst = bf16x8{};
    }
    for (int half = 0; half < 2; ++half) {
      const int r = b_r + half * 16;
      const short* src = (const short*)(B + (int64_t)(k0 + r) * N + n0 + b_c8);
      short* dst = &lds_b[buf][r * BN + b_c8];
      if (n0 + b_c8 < N) *(bf16x8*)dst = *(const bf16x8*)src;
      else *(bf16x8*)dst = bf16x8{};
    }
  }
  __syncthreads();

  for (int k0 = 0; k0 < K; k0 += BK) {
    const int nxt = buf ^ 1;
    if (k0 + BK < K) {
      const int kn = k0 + BK;
      for (int half = 0; half < 2; ++half) {
        const int r = a_r + half * 64;
        const short* src = (const short*)(A + (int64_t)(m0 + r) * K + kn + a_c8);
        short* dst = &lds_a[nxt][r * BK + a_c8];
        if (m0 + r < M) *(bf16x8*)dst = *(const bf16x8*)src;
        else *(bf16x8*)dst = bf16x8{};
      }
      for (int half = 0; half < 2; ++half) {
        const int r = b_r + half * 16;
        const short* src = (const short*)(B + (int64_t)(kn + r) * N + n0 + b_c8);
        short* dst = &lds_b[nxt][r * BN + b_c8];
        if (n0 + b_c8 < N) *(bf16x8*)dst = *(const bf16x8*)src;
        else *(bf16x8*)dst = bf16x8{};
      }
    }
    // compute on buf: two K-substeps of 32... BK == mfma K (32): one step.
    // A frags: rows wr*64 + am*16 + fi, k = fk8..fk8+7
    // B frags: k = fk8.., col wc*64 + bn*16 + fi
#pragma unroll
    for (int am = 0; am < 4; ++am) {
      bf16x8 a_frag;
      const short* ap = &lds_a[buf][(wr * 64 + am * 16 + fi) * BK + fk8];
      a_frag = *(const bf16x8*)ap;
#pragma unroll
      for (int bn = 0; bn < 4; ++bn) {
        bf16x8 b_frag;
        // B fragment: 8 k-values stride BN apart — gathered scalar
        const short* bp = &lds_b[buf][fk8 * BN + wc * 64 + bn * 16 + fi];
#pragma unroll
        for (int j = 0; j < 8; ++j) b_frag[j] = bp[j * BN];
        acc[am][bn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag, b_frag, acc[am][bn], 0, 0, 0);
      }
    }
    buf = nxt;
    __syncthreads();
  }

  // epilogue: bias + activation, store
#pragma unroll
  for (int am = 0; am < 4; ++am) {
#pragma unroll
    for (int bn = 0; bn < 4; ++bn) {
      const int col = n0 + wc * 64 + bn * 16 + fi;
      if (col >= N) continue;
      const float bval = bias ? bias[col] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wr * 64 + am * 16 + (lane >> 4) * 4 + r;
        if (row >= M) continue;
        float v = acc[am][bn][r] + bval;
        if (ACT == 1) v = fmaxf(v, 0.0f);
        if (STORE_F32)
          ((float*)C)[(int64_t)row * ldc + col] = v;
        else
          ((short*)C)[(int64_t)row * ldc + col] = f2bf(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// relu mask backward: g_eff = g * (out > 0)   (bf16, elementwise)
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

// ---------------------------------------------------------------------------
// bias grad: db[N] = sum_m dC[m][n]  (f32 out; one block per 128 cols)
// ---------------------------------------------------------------------------
__global__ void bias_grad_kernel(const short* __restrict__ dC,
                                 float* __restrict__ db, int M, int N) {
  const int col = blockIdx.x * 128 + (threadIdx.x % 128);
  if (col >= N) return;
  const int part = threadIdx.x / 128;  // 2 partitions of rows
  float acc = 0.0f;
  for (int m = part; m < M; m += 2) acc += bf2f(dC[(int64_t)m * N + col]);
  atomicAdd(&db[col], acc);
}

// ---------------------------------------------------------------------------
// wgrad: dW[N,K] += sum_m dC[m][n] * A[m][k]   (dC [M,N], A [M,K] bf16;
// dW f32 — matches the f32 master weights).  64x64 (n,k) tiles, 4 waves as
// 2x2 of 32x32 per-wave tiles (2x2 MFMA frags), M split across SPLITM blocks
// accumulated with atomicAdd (small-K layers underfill the chip otherwise).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void wgrad_kernel(
    const short* __restrict__ dC, const short* __restrict__ A,
    float* __restrict__ dW, int M, int N, int K, int splitm) {
  constexpr int TM = 32;  // m chunk per step
  __shared__ short lds_dc[TM * 64];
  __shared__ short lds_a[TM * 64];

  const int n_tiles_k = (K + 63) / 64;
  const int tile_id = blockIdx.x / splitm;
  const int m_part = blockIdx.x % splitm;
  const int tile_n = tile_id / n_tiles_k;
  const int tile_k = tile_id % n_tiles_k;
  const int n0 = tile_n * 64, k0 = tile_k * 64;

  const int tid = threadIdx.x;
  const int wave = tid / 64, lane = tid % 64;
  const int wr = wave / 2, wc = wave % 2;  // 32x32 wave tile position
  const int fi = lane & 15;
  const int fk8 = (lane >> 4) * 8;

  // m-range of this split
  const int m_chunk = ((M + splitm - 1) / splitm + TM - 1) / TM * TM;
  const int m_begin = m_part * m_chunk;
  const int m_end = min(M, m_begin + m_chunk);

  f32x4 acc[2][2] = {};

  // staging: 256 threads x 8 shorts = 2048 = TM*64
  const int s_r = tid / 8, s_c8 = (tid % 8) * 8;

  for (int m0 = m_begin; m0 < m_end; m0 += TM) {
    {
      const int m = m0 + s_r;
      bf16x8 v{};
      if (m < M && n0 + s_c8 < N)
        v = *(const bf16x8*)(dC + (int64_t)m * N + n0 + s_c8);
      *(bf16x8*)&lds_dc[s_r * 64 + s_c8] = v;
      bf16x8 w{};
      if (m < M && k0 + s_c8 < K)
        w = *(const bf16x8*)(A + (int64_t)m * K + k0 + s_c8);
      *(bf16x8*)&lds_a[s_r * 64 + s_c8] = w;
    }
    __syncthreads();
#pragma unroll
    for (int ks = 0; ks < TM; ks += 32) {  // TM == 32: one step
#pragma unroll
      for (int an = 0; an < 2; ++an) {
        bf16x8 a_frag;  // dC^T fragment: [n][m]
        const int n = wr * 32 + an * 16 + fi;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          a_frag[j] = lds_dc[(ks + fk8 + j) * 64 + n];
#pragma unroll
        for (int bk = 0; bk < 2; ++bk) {
          bf16x8 b_frag;  // A fragment: [m][k]
          const int k = wc * 32 + bk * 16 + fi;
#pragma unroll
          for (int j = 0; j < 8; ++j)
            b_frag[j] = lds_a[(ks + fk8 + j) * 64 + k];
          acc[an][bk] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag, b_frag, acc[an][bk], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int an = 0; an < 2; ++an) {
#pragma unroll
    for (int bk = 0; bk < 2; ++bk) {
      const int k = k0 + wc * 32 + bk * 16 + fi;
      if (k >= K) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int n = n0 + wr * 32 + an * 16 + (lane >> 4) * 4 + r;
        if (n >= N) continue;
        if (splitm > 1)
          atomicAdd(&dW[(int64_t)n * K + k], acc[an][bk][r]);
        else
          dW[(int64_t)n * K + k] = acc[an][bk][r];
      }
    }
  }
}

}  // namespace

// ============================================================ torch bindings

static hipStream_t dcur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

// C = act(A @ B + bias); A [M,K] bf16, B [K,N] bf16 (row-major), bias f32[N]
torch::Tensor gemm_bias_act(torch::Tensor A, torch::Tensor B,
                            torch::Tensor bias, int64_t act,
                            int64_t out_f32) {
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16 && B.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
  const int M = (int)A.size(0), K = (int)A.size(1), N = (int)B.size(1);
  TORCH_CHECK(B.size(0) == K, "gemm shape mismatch");
  TORCH_CHECK(K % BK == 0, "K must be a multiple of 32 (pad)");
  TORCH_CHECK(N % 8 == 0, "N must be a multiple of 8");
  auto C = torch::empty(
      {M, N},
      torch::TensorOptions()
          .dtype(out_f32 ? torch::kFloat32 : torch::kBFloat16)
          .device(A.device()));
  const int grid = ((M + BM - 1) / BM) * ((N + BN - 1) / BN);
  const float* bias_ptr = bias.numel() ? bias.data_ptr<float>() : nullptr;
  hipStream_t st = dcur_stream();
#define PA_GEMM(ACTV, F32V)                                                    \
  hipLaunchKernelGGL((gemm_bias_act_kernel<ACTV, F32V>), dim3(grid), dim3(256),\
                     0, st, (const bf16*)A.data_ptr(), (const bf16*)B.data_ptr(),\
                     bias_ptr, C.data_ptr(), M, N, K, N)
  if (act == 1) { if (out_f32) PA_GEMM(1, true); else PA_GEMM(1, false); }
  else          { if (out_f32) PA_GEMM(0, true); else PA_GEMM(0, false); }
#undef PA_GEMM
  return C;
}

torch::Tensor relu_bwd(torch::Tensor g, torch::Tensor out) {
  auto g_eff = torch::empty_like(g);
  const int64_t n = g.numel();
  TORCH_CHECK(n % 8 == 0);
  const int64_t blocks = (n / 8 + 255) / 256;
  hipLaunchKernelGGL(relu_bwd_kernel, dim3((unsigned)std::min<int64_t>(blocks, 1 << 30)),
                     dim3(256), 0, dcur_stream(), (const short*)g.data_ptr(),
                     (const short*)out.data_ptr(), (short*)g_eff.data_ptr(), n);
  return g_eff;
}

torch::Tensor bias_grad(torch::Tensor dC) {
  const int M = (int)dC.size(0), N = (int)dC.size(1);
  auto db = torch::zeros(
      {N}, torch::TensorOptions().dtype(torch::kFloat32).device(dC.device()));
  hipLaunchKernelGGL(bias_grad_kernel, dim3((N + 127) / 128), dim3(256), 0,
                     dcur_stream(), (const short*)dC.data_ptr(),
                     db.data_ptr<float>(), M, N);
  return db;
}

// dW [N,K] f32 = dC^T @ A
torch::Tensor wgrad(torch::Tensor dC, torch::Tensor A) {
  const int M = (int)dC.size(0), N = (int)dC.size(1), K = (int)A.size(1);
  auto dW = torch::empty(
      {N, K}, torch::TensorOptions().dtype(torch::kFloat32).device(A.device()));
  const int tiles = ((N + 63) / 64) * ((K + 63) / 64);
  int splitm = 1;
  while (tiles * splitm < 512 && splitm < 64 && (M / (splitm * 2)) >= 32)
    splitm *= 2;
  if (splitm > 1) dW.zero_();
  hipLaunchKernelGGL(wgrad_kernel, dim3(tiles * splitm), dim3(256), 0,
                     dcur_stream(), (const short*)dC.data_ptr(),
                     (const short*)A.data_ptr(), dW.data_ptr<float>(), M, N, K,
                     splitm);
  return dW;
}

void init_dense(pybind11::module_& m) {
  m.def("gemm_bias_act", &gemm_bias_act,
        "bf16 MFMA GEMM with fused bias+activation");
  m.def("relu_bwd", &relu_bwd, "g * (out > 0)");
  m.def("bias_grad", &bias_grad, "column-sum bias gradient");
  m.def("wgrad", &wgrad, "dW = dC^T @ A (f32 out)");
}
