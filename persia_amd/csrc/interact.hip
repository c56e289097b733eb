// Fused DLRM dot-interaction kernels (gfx950).
//
// out[b, p(i,j)] = <V[b,i,:], V[b,j,:]>  for the strict lower triangle
// (torch.tril_indices order: i ascending, j<i ascending — models/dlrm.py).
// hipBLASLt runs this as a batched [F,D]x[D,F] GEMM at <10 TF (tiny mats).
//
// v2: per-sample MFMA tiles.  One wave owns one sample; the F x D feature
// matrix is staged in LDS (zero-padded to 16/32 multiples) and the pair
// matrix G = V @ V^T (forward) / dV = sym(G) @ V (backward) is computed
// with mfma_f32_16x16x32_bf16 fragments — 12-16 MFMA ops replace ~1.5k
// VALU MACs per sample (the scalar wave-per-sample version measured ~12%
// VALU efficiency; matrix-shaped work belongs on the matrix cores).
// Fragment maps are the ones verified numerically in csrc/dense.hip:
//   A frag: lane holds A[i = lane&15][k = (lane>>4)*8 + j]
//   B frag: lane holds B[n = lane&15][k = (lane>>4)*8 + j]   (NT form)
//   C frag: lane holds C[row = (lane>>4)*4 + r][col = lane&15]
// Per-wave LDS carve, cross-lane traffic stays inside one wave: no
// __syncthreads anywhere.  Scalar fallback kernels retained for shapes
// whose padded images exceed the 64 KB workgroup LDS limit.
//
// backward: dV[b,i,:] = sum_j G[b,i,j] * V[b,j,:] with G the symmetrized
// pair-gradient matrix (diag 0).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace {

using bf16x8i = __attribute__((ext_vector_type(8))) short;
using bf16x4i = __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16;
using f32x4i = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ float ibf2f(short x) {
  union { float f; unsigned u; } v;
  v.u = ((unsigned)(unsigned short)x) << 16;
  return v.f;
}

__device__ __forceinline__ short if2bf(float f) {
  union { float f; unsigned u; } v;
  v.f = f;
  unsigned lsb = (v.u >> 16) & 1;
  v.u += 0x7fff + lsb;
  return (short)(v.u >> 16);
}

constexpr int ceil16(int x) { return (x + 15) & ~15; }
constexpr int ceil32(int x) { return (x + 31) & ~31; }

// ------------------------------------------------------------- MFMA forward
// one wave per sample: G = V @ V^T on mfma tiles, strict-lower-tri output.
// LDS per wave: Fp x LDK shorts (V image, zero-padded).
__global__ __launch_bounds__(256) void interact_fwd_mfma_kernel(
    const short* __restrict__ V, short* __restrict__ out, int B, int F, int D,
    int P) {
  extern __shared__ short lds[];
  const int waves = blockDim.x / 64;
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const int Fp = ceil16(F), Dk = ceil32(D);
  const int LDK = Dk + 8;
  const int T = Fp / 16;
  short* v = lds + wave * Fp * LDK;
  // zero the image once (covers row/col padding for every sample)
  for (int t = lane; t < Fp * LDK / 8; t += 64) *(bf16x8i*)&v[t * 8] = bf16x8i{};
  const int fi = lane & 15;
  const int fk8 = (lane >> 4) * 8;
  const int d8 = D / 8;
  for (int64_t b = (int64_t)blockIdx.x * waves + wave; b < B;
       b += (int64_t)gridDim.x * waves) {
    const short* src = V + b * F * D;
    for (int t = lane; t < F * d8; t += 64) {
      const int fr = t / d8, c8 = (t % d8) * 8;
      *(bf16x8i*)&v[fr * LDK + c8] = *(const bf16x8i*)&src[fr * D + c8];
    }
    short* dst = out + b * P;
    for (int ti = 0; ti < T; ++ti) {
      for (int tj = 0; tj <= ti; ++tj) {
        f32x4i acc = {};
        for (int k = 0; k < Dk; k += 32) {
          const bf16x8i a = *(const bf16x8i*)&v[(ti * 16 + fi) * LDK + k + fk8];
          const bf16x8i c = *(const bf16x8i*)&v[(tj * 16 + fi) * LDK + k + fk8];
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, c, acc, 0, 0, 0);
        }
        const int j = tj * 16 + fi;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int i = ti * 16 + (lane >> 4) * 4 + r;
          if (i < F && j < i) dst[i * (i - 1) / 2 + j] = if2bf(acc[r]);
        }
      }
    }
  }
}

// ------------------------------------------------------------ MFMA backward
// one wave per sample: dV = sym(G) @ V.
// LDS per wave: A image Fp x LDA (sym pair grads, K = feature axis padded to
// 32) + transposed V image Dp x LDA (so both fragments read K-contiguous).
__global__ __launch_bounds__(256) void interact_bwd_mfma_kernel(
    const short* __restrict__ g, const short* __restrict__ V,
    short* __restrict__ dV, int B, int F, int D, int P) {
  extern __shared__ short lds[];
  const int waves = blockDim.x / 64;
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const int Fp = ceil16(F), Fk = ceil32(F), Dp = ceil16(D);
  const int LDA = Fk + 8;
  short* A = lds + wave * (Fp + Dp) * LDA;
  short* vt = A + Fp * LDA;
  for (int t = lane; t < (Fp + Dp) * LDA / 8; t += 64)
    *(bf16x8i*)&A[t * 8] = bf16x8i{};
  const int Tm = Fp / 16, Tn = Dp / 16;
  const int fi = lane & 15;
  const int fk8 = (lane >> 4) * 8;
  const int d8 = D / 8;
  for (int64_t b = (int64_t)blockIdx.x * waves + wave; b < B;
       b += (int64_t)gridDim.x * waves) {
    // stage V transposed (scalar LDS writes; reads stay b128)
    const short* src = V + b * F * D;
    for (int t = lane; t < F * d8; t += 64) {
      const int fr = t / d8, c8 = (t % d8) * 8;
      const bf16x8i row = *(const bf16x8i*)&src[fr * D + c8];
#pragma unroll
      for (int k = 0; k < 8; ++k) vt[(c8 + k) * LDA + fr] = row[k];
    }
    // pair grads -> symmetric A (diag stays 0); previous sample's entries
    // are overwritten pairwise, padding stays 0 from the one-time zero fill
    const short* gp = g + b * P;
    for (int p = lane; p < P; p += 64) {
      int i = (int)((1.0f + sqrtf(1.0f + 8.0f * (float)p)) * 0.5f);
      while (i * (i - 1) / 2 > p) --i;
      while ((i + 1) * i / 2 <= p) ++i;
      const int j = p - i * (i - 1) / 2;
      const short gv = gp[p];
      A[i * LDA + j] = gv;
      A[j * LDA + i] = gv;
    }
    short* dst = dV + b * F * D;
    for (int ti = 0; ti < Tm; ++ti) {
      for (int td = 0; td < Tn; ++td) {
        f32x4i acc = {};
        for (int k = 0; k < Fk; k += 32) {
          const bf16x8i a = *(const bf16x8i*)&A[(ti * 16 + fi) * LDA + k + fk8];
          const bf16x8i c = *(const bf16x8i*)&vt[(td * 16 + fi) * LDA + k + fk8];
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, c, acc, 0, 0, 0);
        }
        const int d = td * 16 + fi;
        if (d < D) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int i = ti * 16 + (lane >> 4) * 4 + r;
            if (i < F) dst[i * D + d] = if2bf(acc[r]);
          }
        }
      }
    }
  }
}

// ---------------------------------------------- MFMA packed-input variants
// The DLRM flagship path feeds the interaction from TWO sources: the bottom
// MLP output x [B, D] (bf16) and the engine's slot-major sum base
// [S*B, D] (f16, row (f-1)*B + b for feature f>=1).  Staging straight from
// both skips the [B, F, D] cat + permute + f16->bf16 materialization the
// model would otherwise run (and their backward copy cascade).
__global__ __launch_bounds__(256) void interact_fwd_packed_kernel(
    const short* __restrict__ x, const __half* __restrict__ base,
    short* __restrict__ out, int B, int S, int D, int P) {
  extern __shared__ short lds[];
  const int F = S + 1;
  const int waves = blockDim.x / 64;
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const int Fp = ceil16(F), Dk = ceil32(D);
  const int LDK = Dk + 8;
  const int T = Fp / 16;
  short* v = lds + wave * Fp * LDK;
  for (int t = lane; t < Fp * LDK / 8; t += 64) *(bf16x8i*)&v[t * 8] = bf16x8i{};
  const int fi = lane & 15;
  const int fk8 = (lane >> 4) * 8;
  const int d8 = D / 8;
  for (int64_t b = (int64_t)blockIdx.x * waves + wave; b < B;
       b += (int64_t)gridDim.x * waves) {
    for (int t = lane; t < F * d8; t += 64) {
      const int fr = t / d8, c8 = (t % d8) * 8;
      bf16x8i val;
      if (fr == 0) {
        val = *(const bf16x8i*)&x[b * D + c8];
      } else {
        const __half* src = base + ((int64_t)(fr - 1) * B + b) * D + c8;
#pragma unroll
        for (int k = 0; k < 8; ++k) val[k] = if2bf(__half2float(src[k]));
      }
      *(bf16x8i*)&v[fr * LDK + c8] = val;
    }
    short* dst = out + b * P;
    for (int ti = 0; ti < T; ++ti) {
      for (int tj = 0; tj <= ti; ++tj) {
        f32x4i acc = {};
        for (int k = 0; k < Dk; k += 32) {
          const bf16x8i a = *(const bf16x8i*)&v[(ti * 16 + fi) * LDK + k + fk8];
          const bf16x8i c = *(const bf16x8i*)&v[(tj * 16 + fi) * LDK + k + fk8];
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, c, acc, 0, 0, 0);
        }
        const int j = tj * 16 + fi;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int i = ti * 16 + (lane >> 4) * 4 + r;
          if (i < F && j < i) dst[i * (i - 1) / 2 + j] = if2bf(acc[r]);
        }
      }
    }
  }
}

__global__ __launch_bounds__(256) void interact_bwd_packed_kernel(
    const short* __restrict__ g, const short* __restrict__ x,
    const __half* __restrict__ base, short* __restrict__ dx,
    __half* __restrict__ dbase, int B, int S, int D, int P) {
  extern __shared__ short lds[];
  const int F = S + 1;
  const int waves = blockDim.x / 64;
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const int Fp = ceil16(F), Fk = ceil32(F), Dp = ceil16(D);
  const int LDA = Fk + 8;   // G image rows i, cols f'
  const int LDV = Dp + 8;   // V image rows f', cols d (LINEAR: no transpose)
  short* A = lds + wave * (Fp * LDA + Fk * LDV);
  short* v = A + Fp * LDA;
  for (int t = lane; t < (Fp * LDA + Fk * LDV) / 8; t += 64)
    *(bf16x8i*)&A[t * 8] = bf16x8i{};
  const int Tm = Fp / 16, Tn = Dp / 16;
  const int fi = lane & 15;
  const int fk8 = (lane >> 4) * 8;
  const int d8 = D / 8;
  // V^T fragments come from ds_read_b64_tr_b16 on the LINEAR V image (the
  // round-1 transposed scalar staging was the kernel's measured LDS
  // bottleneck).  Addressing per the measured semantics (tools/trprobe +
  // dense.hip wgrad_tr_kernel): lane m of a 16-lane group addresses the
  // slot (row_base + m/4)*LDV + col_base + (m%4)*4 shorts; the group's
  // slots form a 4x16 row-major block and lane m receives its column —
  // with row_base = k + 8*(lane>>4) the fragment lands in the nominal
  // mfma_16x16x32 k-layout, consistent with the row-read A fragment.
  const int t_ro = ((lane & 15) >> 2);
  const int t_co = (lane & 3) * 4;
  auto vtr = [&](int kb, int td) {
    const short* b0 = &v[(kb + fk8 + t_ro) * LDV + td * 16 + t_co];
    bf16x4i lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (__attribute__((address_space(3))) bf16x4i*)(void*)const_cast<short*>(b0));
    bf16x4i hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (__attribute__((address_space(3))) bf16x4i*)(void*)const_cast<short*>(b0 + 4 * LDV));
    bf16x8i r;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      r[j] = ((const short*)&lo)[j];
      r[4 + j] = ((const short*)&hi)[j];
    }
    return r;
  };
  for (int64_t b = (int64_t)blockIdx.x * waves + wave; b < B;
       b += (int64_t)gridDim.x * waves) {
    for (int t = lane; t < F * d8; t += 64) {
      const int fr = t / d8, c8 = (t % d8) * 8;
      bf16x8i row;
      if (fr == 0) {
        row = *(const bf16x8i*)&x[b * D + c8];
      } else {
        const __half* src = base + ((int64_t)(fr - 1) * B + b) * D + c8;
#pragma unroll
        for (int k = 0; k < 8; ++k) row[k] = if2bf(__half2float(src[k]));
      }
      *(bf16x8i*)&v[fr * LDV + c8] = row;
    }
    const short* gp = g + b * P;
    for (int p = lane; p < P; p += 64) {
      int i = (int)((1.0f + sqrtf(1.0f + 8.0f * (float)p)) * 0.5f);
      while (i * (i - 1) / 2 > p) --i;
      while ((i + 1) * i / 2 <= p) ++i;
      const int j = p - i * (i - 1) / 2;
      const short gv = gp[p];
      A[i * LDA + j] = gv;
      A[j * LDA + i] = gv;
    }
    for (int ti = 0; ti < Tm; ++ti) {
      for (int td = 0; td < Tn; ++td) {
        f32x4i acc = {};
        for (int k = 0; k < Fk; k += 32) {
          const bf16x8i a = *(const bf16x8i*)&A[(ti * 16 + fi) * LDA + k + fk8];
          const bf16x8i c = vtr(k, td);
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, c, acc, 0, 0, 0);
        }
        const int d = td * 16 + fi;
        if (d < D) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int i = ti * 16 + (lane >> 4) * 4 + r;
            if (i == 0) {
              dx[b * D + d] = if2bf(acc[r]);
            } else if (i < F) {
              dbase[((int64_t)(i - 1) * B + b) * D + d] = __float2half(acc[r]);
            }
          }
        }
      }
    }
  }
}

// ------------------------------------------------- scalar fallback (legacy)
// one wave per sample; V_b staged in LDS; lanes split the P pairs
__global__ __launch_bounds__(256) void interact_fwd_kernel(
    const short* __restrict__ V, short* __restrict__ out, int B, int F, int D,
    int P) {
  extern __shared__ short lds[];
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const int LD = D + 8;  // +8-short row pad: conflict-free same-column reads
  short* v = lds + wave * F * LD;
  // uniform trip count across all waves of the block (barriers inside)
  const int64_t stride = (int64_t)gridDim.x * 4;
  const int64_t rounds = ((int64_t)B + stride - 1) / stride;
  for (int64_t r = 0; r < rounds; ++r) {
    const int64_t b = r * stride + (int64_t)blockIdx.x * 4 + wave;
    if (b < B) {
      const short* src = V + b * F * D;
      const int d8 = D / 8;
      for (int t = lane; t < F * d8; t += 64) {
        const int fr = t / d8, c8 = (t % d8) * 8;
        *(bf16x8i*)&v[fr * LD + c8] = *(const bf16x8i*)&src[fr * D + c8];
      }
    }
    __syncthreads();
    if (b < B) {
      for (int p = lane; p < P; p += 64) {
        // invert p -> (i, j): i = largest with i*(i-1)/2 <= p
        int i = (int)((1.0f + sqrtf(1.0f + 8.0f * (float)p)) * 0.5f);
        while (i * (i - 1) / 2 > p) --i;
        while ((i + 1) * i / 2 <= p) ++i;
        const int j = p - i * (i - 1) / 2;
        const short* vi = &v[i * LD];
        const short* vj = &v[j * LD];
        float acc = 0.0f;
        for (int d = 0; d < D; d += 8) {
          bf16x8i a = *(const bf16x8i*)&vi[d];
          bf16x8i c = *(const bf16x8i*)&vj[d];
#pragma unroll
          for (int k = 0; k < 8; ++k) acc += ibf2f(a[k]) * ibf2f(c[k]);
        }
        out[b * P + p] = if2bf(acc);
      }
    }
    __syncthreads();
  }
}

// one wave per sample: dV = G_sym @ V (G from pair grads, diag 0)
__global__ __launch_bounds__(256) void interact_bwd_kernel(
    const short* __restrict__ g, const short* __restrict__ V,
    short* __restrict__ dV, int B, int F, int D, int P) {
  extern __shared__ short lds[];
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const int LD = D + 8;
  short* v = lds + wave * (F * LD + F * F);
  short* G = v + F * LD;
  const int64_t stride = (int64_t)gridDim.x * 4;
  const int64_t rounds = ((int64_t)B + stride - 1) / stride;
  for (int64_t r = 0; r < rounds; ++r) {
    const int64_t b = r * stride + (int64_t)blockIdx.x * 4 + wave;
    if (b < B) {
      const short* src = V + b * F * D;
      const int d8l = D / 8;
      for (int t = lane; t < F * d8l; t += 64) {
        const int fr = t / d8l, c8 = (t % d8l) * 8;
        *(bf16x8i*)&v[fr * LD + c8] = *(const bf16x8i*)&src[fr * D + c8];
      }
      for (int t = lane; t < F * F; t += 64) G[t] = 0;
    }
    __syncthreads();
    if (b < B) {
      const short* gp = g + b * P;
      for (int p = lane; p < P; p += 64) {
        int i = (int)((1.0f + sqrtf(1.0f + 8.0f * (float)p)) * 0.5f);
        while (i * (i - 1) / 2 > p) --i;
        while ((i + 1) * i / 2 <= p) ++i;
        const int j = p - i * (i - 1) / 2;
        const short gv = gp[p];
        G[i * F + j] = gv;
        G[j * F + i] = gv;
      }
    }
    __syncthreads();
    if (b < B) {
      // dV[i][d] = sum_j G[i][j] * v[j][d]; lanes tile (i, d8)
      const int d8 = D / 8;
      short* dst = dV + b * F * D;
      for (int t = lane; t < F * d8; t += 64) {
        const int i = t / d8, dd = (t % d8) * 8;
        float acc[8] = {};
        for (int j = 0; j < F; ++j) {
          const float gij = ibf2f(G[i * F + j]);
          if (gij != 0.0f) {
            const short* vj = &v[j * LD + dd];
#pragma unroll
            for (int k = 0; k < 8; ++k) acc[k] += gij * ibf2f(vj[k]);
          }
        }
#pragma unroll
        for (int k = 0; k < 8; ++k) dst[i * D + dd + k] = if2bf(acc[k]);
      }
    }
    __syncthreads();
  }
}

// pick the widest block whose per-wave LDS carve fits the 64 KB workgroup
// limit; 0 waves = doesn't fit at all
inline int pick_waves(int per_wave_shorts) {
  const int bytes = per_wave_shorts * 2;
  for (int w = 4; w >= 1; w >>= 1)
    if (w * bytes <= 65536) return w;
  return 0;
}

}  // namespace

static hipStream_t icur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

torch::Tensor interact_fwd(torch::Tensor V) {
  TORCH_CHECK(V.scalar_type() == torch::kBFloat16 && V.is_contiguous());
  const int B = (int)V.size(0), F = (int)V.size(1), D = (int)V.size(2);
  TORCH_CHECK(D % 8 == 0);
  const int P = F * (F - 1) / 2;
  auto out = torch::empty(
      {B, P}, torch::TensorOptions().dtype(torch::kBFloat16).device(V.device()));
  const int Fp = ceil16(F), Dk = ceil32(D);
  const int mfma_shorts = Fp * (Dk + 8);
  const int waves = pick_waves(mfma_shorts);
  if (waves > 0) {
    const int grid = std::min((B + waves - 1) / waves, 8192);
    hipLaunchKernelGGL(interact_fwd_mfma_kernel, dim3(grid),
                       dim3(waves * 64), waves * mfma_shorts * 2,
                       icur_stream(), (const short*)V.data_ptr(),
                       (short*)out.data_ptr(), B, F, D, P);
    return out;
  }
  const int lds = 4 * F * (D + 8) * 2;
  TORCH_CHECK(lds <= 65536, "interaction tile exceeds LDS");
  const int grid = std::min((B + 3) / 4, 8192);
  hipLaunchKernelGGL(interact_fwd_kernel, dim3(grid), dim3(256), lds,
                     icur_stream(), (const short*)V.data_ptr(),
                     (short*)out.data_ptr(), B, F, D, P);
  return out;
}

torch::Tensor interact_bwd(torch::Tensor g, torch::Tensor V) {
  const int B = (int)V.size(0), F = (int)V.size(1), D = (int)V.size(2);
  const int P = F * (F - 1) / 2;
  auto dV = torch::empty_like(V);
  const int Fp = ceil16(F), Fk = ceil32(F), Dp = ceil16(D);
  const int mfma_shorts = (Fp + Dp) * (Fk + 8);
  const int waves = pick_waves(mfma_shorts);
  if (waves > 0) {
    const int grid = std::min((B + waves - 1) / waves, 8192);
    hipLaunchKernelGGL(interact_bwd_mfma_kernel, dim3(grid),
                       dim3(waves * 64), waves * mfma_shorts * 2,
                       icur_stream(), (const short*)g.data_ptr(),
                       (const short*)V.data_ptr(), (short*)dV.data_ptr(), B, F,
                       D, P);
    return dV;
  }
  const int lds = 4 * (F * (D + 8) * 2 + 2 * F * F);
  TORCH_CHECK(lds <= 65536, "interaction tile exceeds LDS");
  const int grid = std::min((B + 3) / 4, 8192);
  hipLaunchKernelGGL(interact_bwd_kernel, dim3(grid), dim3(256), lds,
                     icur_stream(), (const short*)g.data_ptr(),
                     (const short*)V.data_ptr(), (short*)dV.data_ptr(), B, F,
                     D, P);
  return dV;
}

torch::Tensor interact_fwd_packed(torch::Tensor x, torch::Tensor base) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(base.scalar_type() == torch::kFloat16 && base.is_contiguous());
  const int B = (int)x.size(0), D = (int)x.size(1);
  const int S = (int)(base.size(0) / B);
  const int F = S + 1;
  TORCH_CHECK(D % 8 == 0 && base.size(1) == D && base.size(0) == (int64_t)S * B);
  const int P = F * (F - 1) / 2;
  auto out = torch::empty(
      {B, P}, torch::TensorOptions().dtype(torch::kBFloat16).device(x.device()));
  const int Fp = ceil16(F), Dk = ceil32(D);
  const int mfma_shorts = Fp * (Dk + 8);
  const int waves = pick_waves(mfma_shorts);
  TORCH_CHECK(waves > 0, "interact_fwd_packed: tile exceeds LDS");
  const int grid = std::min((B + waves - 1) / waves, 8192);
  hipLaunchKernelGGL(interact_fwd_packed_kernel, dim3(grid), dim3(waves * 64),
                     waves * mfma_shorts * 2, icur_stream(),
                     (const short*)x.data_ptr(),
                     (const __half*)base.data_ptr<at::Half>(),
                     (short*)out.data_ptr(), B, S, D, P);
  return out;
}

std::vector<torch::Tensor> interact_bwd_packed(torch::Tensor g, torch::Tensor x,
                                               torch::Tensor base) {
  const int B = (int)x.size(0), D = (int)x.size(1);
  const int S = (int)(base.size(0) / B);
  const int F = S + 1;
  const int P = F * (F - 1) / 2;
  auto dx = torch::empty_like(x);
  auto dbase = torch::empty_like(base);
  const int Fp = ceil16(F), Fk = ceil32(F), Dp = ceil16(D);
  const int mfma_shorts = Fp * (Fk + 8) + Fk * (Dp + 8);
  const int waves = pick_waves(mfma_shorts);
  TORCH_CHECK(waves > 0, "interact_bwd_packed: tile exceeds LDS");
  const int grid = std::min((B + waves - 1) / waves, 8192);
  auto gc = g.contiguous();
  hipLaunchKernelGGL(interact_bwd_packed_kernel, dim3(grid), dim3(waves * 64),
                     waves * mfma_shorts * 2, icur_stream(),
                     (const short*)gc.data_ptr(),
                     (const short*)x.data_ptr(),
                     (const __half*)base.data_ptr<at::Half>(),
                     (short*)dx.data_ptr(), (__half*)dbase.data_ptr<at::Half>(),
                     B, S, D, P);
  return {dx, dbase};
}

// packed variants have no scalar fallback: MFMA images must fit
bool interact_packed_feasible(int64_t F, int64_t D) {
  if (D % 8 != 0) return false;
  const int Fp = ceil16((int)F), Fk = ceil32((int)F), Dp = ceil16((int)D),
            Dk = ceil32((int)D);
  return pick_waves(Fp * (Dk + 8)) > 0 &&
         pick_waves(Fp * (Fk + 8) + Fk * (Dp + 8)) > 0;
}

bool interact_feasible(int64_t F, int64_t D) {
  if (D % 8 != 0) return false;
  const int Fp = ceil16((int)F), Fk = ceil32((int)F), Dp = ceil16((int)D),
            Dk = ceil32((int)D);
  const bool fwd_ok = pick_waves(Fp * (Dk + 8)) > 0 ||
                      4 * (int)F * ((int)D + 8) * 2 <= 65536;
  const bool bwd_ok =
      pick_waves((Fp + Dp) * (Fk + 8)) > 0 ||
      4 * ((int)F * ((int)D + 8) * 2 + 2 * (int)F * (int)F) <= 65536;
  return fwd_ok && bwd_ok;
}

void init_interact(pybind11::module_& m) {
  m.def("interact_fwd", &interact_fwd, "DLRM pairwise dot interaction");
  m.def("interact_bwd", &interact_bwd, "interaction backward (dV)");
  m.def("interact_feasible", &interact_feasible,
        "true if a fused interaction kernel exists for (F, D)");
  m.def("interact_packed_feasible", &interact_packed_feasible,
        "true if the packed-input MFMA interaction fits (F, D)");
  m.def("interact_fwd_packed", &interact_fwd_packed,
        "interaction from (x bf16 [B,D], slot-major f16 base) directly");
  m.def("interact_bwd_packed", &interact_bwd_packed,
        "packed interaction backward -> (dx, dbase f16)");
}
