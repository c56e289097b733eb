// Fused DLRM dot-interaction kernels (gfx950).
//
// out[b, p(i,j)] = <V[b,i,:], V[b,j,:]>  for the strict lower triangle
// (torch.tril_indices order: i ascending, j<i ascending — models/dlrm.py).
// hipBLASLt runs this as a batched [F,D]x[D,F] GEMM at <10 TF (tiny mats);
// here one wave owns one sample with V staged in LDS — the op is
// memory-bound on reading V once (B*F*D bf16) instead of GEMM-shaped.
//
// backward: dV[b,i,:] = sum_j G[b,i,j] * V[b,j,:] with G the symmetrized
// pair-gradient matrix (diag 0).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace {

using bf16x8i = __attribute__((ext_vector_type(8))) short;

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
  // per-wave carve: V (F*LD shorts) + G (F*F bf16 — the incoming pair grads
  // are bf16, so storing G as bf16 loses nothing; halves LDS so F=65/dim-8
  // configs fit)
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
  const int lds = 4 * F * (D + 8) * 2;
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
  const int lds = 4 * (F * (D + 8) * 2 + 2 * F * F);
  TORCH_CHECK(lds <= 65536, "interaction tile exceeds LDS");
  const int grid = std::min((B + 3) / 4, 8192);
  hipLaunchKernelGGL(interact_bwd_kernel, dim3(grid), dim3(256), lds,
                     icur_stream(), (const short*)g.data_ptr(),
                     (const short*)V.data_ptr(), (short*)dV.data_ptr(), B, F,
                     D, P);
  return dV;
}

void init_interact(pybind11::module_& m) {
  m.def("interact_fwd", &interact_fwd, "DLRM pairwise dot interaction");
  m.def("interact_bwd", &interact_bwd, "interaction backward (dV)");
}
