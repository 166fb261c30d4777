// CDNA4 (gfx950 / MI355X) HIP kernels for distributed K-FAC.
//
// Written natively for MI355X: 64-wide wavefronts, MFMA matrix cores
// (bf16 in / fp32 accumulate), LDS-tiled staging with conflict-free
// padded layouts.  No CUDA compat paths, no hipify.
//
// Kernels:
//   * syrk_bf16_kernel    -- the K-FAC factor product
//       F = (s*[X|1])^T (s*[X|1]) / denom  for X[rows, d] bf16,
//     fused bias ones-column, split-K atomics into an fp32 workspace.
//     Replaces the reference's cat(ones) + div_ + aT@a GEMM chain
//     (reference: kfac/utils.py:86-103) and its fp16 tensor-core side
//     path tcmm_gemm_ex (reference: packages/tcmm/src/tcmm_kernel.cu:125-157).
//   * syrk_epilogue_kernel -- fused scale + running-average
//       out = (1-decay)*out + decay*alpha*tmp
//     (reference: update_running_avg, kfac/utils.py:66-71).
//   * eigen_scale_kernel  -- V /= (dG dA^T + damping), the rank-1
//     denominator of the implicit-eigen preconditioner, no dG*dA^T
//     materialization (reference: kfac/kfac_preconditioner_eigen.py:142).
//   * im2col_kernel       -- conv patch rows (B*oh*ow, C*kh*kw) in bf16,
//     matching the reference layout (reference: kfac/utils.py:33-54).

#include <torch/extension.h>

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <c10/hip/HIPStream.h>

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));    \
  } while (0)

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) short;
using ushort8 = __attribute__((ext_vector_type(8))) unsigned short;

namespace {

// ---------------------------------------------------------------------------
// eigen_scale: v[i, j] /= (dG[i] * dA[j] + damping)
// ---------------------------------------------------------------------------
__global__ void eigen_scale_kernel(float* __restrict__ v,
                                   const float* __restrict__ dG,
                                   const float* __restrict__ dA,
                                   float damping, long ng, long na) {
  const long total = ng * na;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long i = idx / na;
    const long j = idx - i * na;
    v[idx] /= (dG[i] * dA[j] + damping);
  }
}

// batched: v[b, ng, na] /= (dG[b, ng] dA[b, na]^T + damping)
__global__ void eigen_scale_batched_kernel(float* __restrict__ v,
                                           const float* __restrict__ dG,
                                           const float* __restrict__ dA,
                                           float damping, long nb, long ng,
                                           long na) {
  const long per = ng * na;
  const long total = nb * per;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long b = idx / per;
    const long r = idx - b * per;
    const long i = r / na;
    const long j = r - i * na;
    v[idx] /= (dG[b * ng + i] * dA[b * na + j] + damping);
  }
}

// ---------------------------------------------------------------------------
// SYRK factor kernel: Ctmp += [X|1]^T [X|1]  (raw accumulation; the
// scale s^2/denom and the running average land in the epilogue).
// X is bf16 [rows, d] row-major (ldx halves between rows).
//
// Geometry: 128x128 output tile per workgroup, 4 waves in 2x2, each wave
// a 64x64 sub-tile of 4x4 mfma_f32_16x16x32_bf16 fragments; K-step 32.
// LDS tiles are stored TRANSPOSED as [col][k] with a 56-half (112 B) row
// stride: every ds_read_b128 stays 16-B aligned, the contiguous 8-lane
// write groups see banks 28*c mod 32 (distinct for c=0..7) and the
// 16-lane read groups see 28*c mod 64 (distinct for c=0..15) -- both
// conflict-free.
// Only upper-triangle tiles are computed; the epilogue mirrors each
// off-diagonal tile with a transposed atomic add.
// Split-K across workgroups accumulates with fp32 atomics so small-d
// factors (d=27..512, rows up to ~400k for ResNet-50 im2col) still fill
// all 256 CUs.
// ---------------------------------------------------------------------------
constexpr int BT = 128;         // output tile edge
constexpr int BK = 32;          // K-step (mfma 16x16x32)
constexpr int LDS_STRIDE = 56;  // halves per LDS [col][k] row
constexpr unsigned short BF16_ONE = 0x3F80;

__device__ __forceinline__ void stage_tile(
    const unsigned short* __restrict__ X, unsigned short* lds, int c0,
    long k0, long kend, long ldx, int d, int bias_col) {
  // Stage X[k0:k0+32, c0:c0+128] as lds[col][k]; out-of-range k -> 0,
  // col == bias_col -> 1.0, col past the bias column -> 0.
  const int t = threadIdx.x;
#pragma unroll
  for (int rep = 0; rep < 2; ++rep) {
    const int a = t + rep * 256;       // 0..511
    const int c = a & 127;             // column within tile
    const int kg = a >> 7;             // k-group (0..3), 8 k each
    const int gc = c0 + c;
    unsigned short vals[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const long gk = k0 + kg * 8 + e;
      unsigned short v = 0;
      if (gk < kend) {
        if (gc < d) {
          v = X[gk * ldx + gc];
        } else if (gc == bias_col) {
          v = BF16_ONE;
        }
      }
      vals[e] = v;
    }
    *reinterpret_cast<ushort8*>(&lds[c * LDS_STRIDE + kg * 8]) =
        *reinterpret_cast<const ushort8*>(vals);
  }
}

__global__ __launch_bounds__(256) void syrk_bf16_kernel(
    const unsigned short* __restrict__ X, float* __restrict__ Ctmp,
    long rows, int d, long ldx, int n, int bias, int nsplit, long ksplit,
    int ntiles, int nslabs) {
  // blockIdx.x -> (upper-triangle tile pair, K split); splits spread
  // their atomics over nslabs separate accumulation slabs so hundreds
  // of split-K blocks do not hammer one tiny n x n output (d <= 128
  // factors showed ~4x atomic-contention slowdowns)
  const int tp = blockIdx.x / nsplit;
  const int split = blockIdx.x - tp * nsplit;
  Ctmp += (long)(split % nslabs) * n * n;
  // tile pair tp -> (ti, tj) with ti <= tj, row-major over the triangle
  int ti = 0, rem = tp;
  while (rem >= ntiles - ti) {
    rem -= ntiles - ti;
    ++ti;
  }
  const int tj = ti + rem;
  const bool diag = (ti == tj);

  const long k_begin = (long)split * ksplit;
  long k_end = k_begin + ksplit;
  if (k_end > rows) k_end = rows;
  if (k_begin >= k_end) return;

  const int bias_col = bias ? d : -1;

  __shared__ unsigned short ldsA[BT * LDS_STRIDE];
  __shared__ unsigned short ldsB[BT * LDS_STRIDE];
  unsigned short* tileA = ldsA;                    // C-rows:  cols ti*128..
  unsigned short* tileB = diag ? ldsA : ldsB;      // C-cols:  cols tj*128..

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wr = wave >> 1;          // wave row band (0..1) * 64
  const int wc = wave & 1;           // wave col band (0..1) * 64
  const int l15 = lane & 15;
  const int l4 = lane >> 4;          // 0..3

  f32x4 acc[4][4] = {};

  for (long k0 = k_begin; k0 < k_end; k0 += BK) {
    stage_tile(X, tileA, ti * BT, k0, k_end, ldx, d, bias_col);
    if (!diag) {
      stage_tile(X, tileB, tj * BT, k0, k_end, ldx, d, bias_col);
    }
    __syncthreads();

    bf16x8 afrag[4], bfrag[4];
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) {
      const int m_local = wr * 64 + mf * 16 + l15;
      afrag[mf] = *reinterpret_cast<const bf16x8*>(
          &tileA[m_local * LDS_STRIDE + l4 * 8]);
    }
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int n_local = wc * 64 + nf * 16 + l15;
      bfrag[nf] = *reinterpret_cast<const bf16x8*>(
          &tileB[n_local * LDS_STRIDE + l4 * 8]);
    }
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[mf], bfrag[nf], acc[mf][nf], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // epilogue: atomic accumulate (split-K), mirroring off-diagonal tiles
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int i = ti * BT + wr * 64 + mf * 16 + l4 * 4 + r;
        const int j = tj * BT + wc * 64 + nf * 16 + l15;
        if (i < n && j < n) {
          const float val = acc[mf][nf][r];
          atomicAdd(&Ctmp[(long)i * n + j], val);
          if (!diag) {
            atomicAdd(&Ctmp[(long)j * n + i], val);
          }
        }
      }
    }
  }
}

__global__ void syrk_epilogue_kernel(float* __restrict__ out,
                                     const float* __restrict__ tmp,
                                     float alpha, float decay, long total,
                                     int nslabs) {
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int s = 0; s < nslabs; ++s) {
      acc += tmp[s * total + idx];
    }
    const float fresh = acc * alpha;
    out[idx] = (decay < 0.f) ? fresh
                             : (1.f - decay) * out[idx] + decay * fresh;
  }
}

// ---------------------------------------------------------------------------
// im2col: (B, C, H, W) -> (B*oh*ow, C*kh*kw) bf16 patch rows
// ---------------------------------------------------------------------------
template <typename scalar_t>
__device__ __forceinline__ unsigned short to_bf16_bits(scalar_t v);

template <>
__device__ __forceinline__ unsigned short to_bf16_bits<float>(float v) {
  __hip_bfloat16 b = __float2bfloat16(v);
  return *reinterpret_cast<unsigned short*>(&b);
}

template <>
__device__ __forceinline__ unsigned short to_bf16_bits<unsigned short>(
    unsigned short v) {
  return v;
}

template <typename scalar_t>
__global__ void im2col_kernel(const scalar_t* __restrict__ x,
                              unsigned short* __restrict__ out, int B, int C,
                              int H, int W, int kh, int kw, int sh, int sw,
                              int ph, int pw, int dh, int dw, int oh,
                              int ow) {
  const long ckk = (long)C * kh * kw;
  const long total = (long)B * oh * ow * ckk;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / ckk;
    const int col = (int)(idx - row * ckk);
    const int b = (int)(row / ((long)oh * ow));
    const int ij = (int)(row - (long)b * oh * ow);
    const int i = ij / ow;
    const int j = ij - i * ow;
    const int c = col / (kh * kw);
    const int uv = col - c * (kh * kw);
    const int u = uv / kw;
    const int v = uv - u * kw;
    const int hi = i * sh - ph + u * dh;
    const int wj = j * sw - pw + v * dw;
    unsigned short o = 0;
    if (hi >= 0 && hi < H && wj >= 0 && wj < W) {
      o = to_bf16_bits<scalar_t>(
          x[(((long)b * C + c) * H + hi) * W + wj]);
    }
    out[idx] = o;
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// host entry points
// ---------------------------------------------------------------------------

void eigen_scale_entry(torch::Tensor v, torch::Tensor dG, torch::Tensor dA,
                       double damping) {
  TORCH_CHECK(v.is_cuda() && dG.is_cuda() && dA.is_cuda(),
              "eigen_scale_: tensors must be on GPU");
  TORCH_CHECK(v.is_contiguous() && dG.is_contiguous() && dA.is_contiguous(),
              "eigen_scale_: tensors must be contiguous");
  TORCH_CHECK(v.scalar_type() == at::kFloat &&
                  dG.scalar_type() == at::kFloat &&
                  dA.scalar_type() == at::kFloat,
              "eigen_scale_: fp32 only");
  const long ng = v.size(0), na = v.size(1);
  TORCH_CHECK(dG.numel() == ng && dA.numel() == na, "eigen_scale_: shape");
  auto stream = c10::hip::getCurrentHIPStream();
  const long total = ng * na;
  const int block = 256;
  const int grid = (int)std::min<long>((total + block - 1) / block, 2048);
  eigen_scale_kernel<<<grid, block, 0, stream.stream()>>>(
      v.data_ptr<float>(), dG.data_ptr<float>(), dA.data_ptr<float>(),
      (float)damping, ng, na);
  HIP_CHECK(hipGetLastError());
}

void eigen_scale_batched_entry(torch::Tensor v, torch::Tensor dG,
                               torch::Tensor dA, double damping) {
  TORCH_CHECK(v.is_cuda() && v.dim() == 3 && v.is_contiguous() &&
                  dG.is_contiguous() && dA.is_contiguous(),
              "eigen_scale_batched_: contiguous GPU tensors required");
  TORCH_CHECK(v.scalar_type() == at::kFloat &&
                  dG.scalar_type() == at::kFloat &&
                  dA.scalar_type() == at::kFloat,
              "eigen_scale_batched_: fp32 only");
  const long nb = v.size(0), ng = v.size(1), na = v.size(2);
  TORCH_CHECK(dG.numel() == nb * ng && dA.numel() == nb * na,
              "eigen_scale_batched_: shape mismatch");
  auto stream = c10::hip::getCurrentHIPStream();
  const long total = nb * ng * na;
  const int block = 256;
  const int grid = (int)std::min<long>((total + block - 1) / block, 4096);
  eigen_scale_batched_kernel<<<grid, block, 0, stream.stream()>>>(
      v.data_ptr<float>(), dG.data_ptr<float>(), dA.data_ptr<float>(),
      (float)damping, nb, ng, na);
  HIP_CHECK(hipGetLastError());
}

torch::Tensor syrk_factor_entry(torch::Tensor x, torch::Tensor out,
                                double row_scale, double denom, bool bias,
                                double decay) {
  TORCH_CHECK(x.is_cuda() && out.is_cuda(), "syrk_factor_: GPU tensors only");
  TORCH_CHECK(x.dim() == 2 && x.is_contiguous(),
              "syrk_factor_: x must be 2-D contiguous");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16,
              "syrk_factor_: x must be bf16 (cast capture to bf16 first)");
  TORCH_CHECK(out.scalar_type() == at::kFloat && out.is_contiguous(),
              "syrk_factor_: out must be fp32 contiguous");
  const long rows = x.size(0);
  const int d = (int)x.size(1);
  const int n = d + (bias ? 1 : 0);
  TORCH_CHECK(out.size(0) == n && out.size(1) == n,
              "syrk_factor_: out shape mismatch, expected ", n, "x", n);
  TORCH_CHECK(rows > 0, "syrk_factor_: empty input");

  auto stream = c10::hip::getCurrentHIPStream();

  const int ntiles = (n + BT - 1) / BT;
  const long npairs = (long)ntiles * (ntiles + 1) / 2;
  // split K so the grid covers the chip (256 CUs want >= ~512 blocks)
  long nsplit = (640 + npairs - 1) / npairs;
  const long max_split = (rows + 1023) / 1024;
  if (nsplit > max_split) nsplit = max_split;
  if (nsplit < 1) nsplit = 1;
  long ksplit = (rows + nsplit - 1) / nsplit;
  ksplit = (ksplit + BK - 1) / BK * BK;
  // spread split-K atomics over up to 16 accumulation slabs (bounded
  // extra memory: 16 * n^2 floats, n <= a few K)
  const int nslabs = (int)std::min<long>(nsplit, 16);

  auto tmp = at::zeros({(long)nslabs, (long)n, (long)n}, out.options());
  const long grid = npairs * nsplit;
  syrk_bf16_kernel<<<(int)grid, 256, 0, stream.stream()>>>(
      reinterpret_cast<const unsigned short*>(x.data_ptr<at::BFloat16>()),
      tmp.data_ptr<float>(), rows, d, x.stride(0), n, bias ? 1 : 0,
      (int)nsplit, ksplit, ntiles, nslabs);
  HIP_CHECK(hipGetLastError());

  const float alpha = (float)(row_scale * row_scale / denom);
  const long total = (long)n * n;
  const int grid2 = (int)std::min<long>((total + 255) / 256, 2048);
  syrk_epilogue_kernel<<<grid2, 256, 0, stream.stream()>>>(
      out.data_ptr<float>(), tmp.data_ptr<float>(), alpha, (float)decay,
      total, nslabs);
  HIP_CHECK(hipGetLastError());
  return out;
}

torch::Tensor im2col_entry(torch::Tensor x, long kh, long kw, long sh,
                           long sw, long ph, long pw, long dh, long dw) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous(),
              "im2col: x must be 4-D contiguous GPU tensor");
  const int B = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3);
  const int oh = (H + 2 * (int)ph - (int)dh * ((int)kh - 1) - 1) / (int)sh + 1;
  const int ow = (W + 2 * (int)pw - (int)dw * ((int)kw - 1) - 1) / (int)sw + 1;
  TORCH_CHECK(oh > 0 && ow > 0, "im2col: empty output");
  auto out = at::empty({(long)B * oh * ow, (long)C * kh * kw},
                       x.options().dtype(at::kBFloat16));
  auto stream = c10::hip::getCurrentHIPStream();
  const long total = out.numel();
  const int block = 256;
  const int grid = (int)std::min<long>((total + block - 1) / block, 4096);
  if (x.scalar_type() == at::kFloat) {
    im2col_kernel<float><<<grid, block, 0, stream.stream()>>>(
        x.data_ptr<float>(),
        reinterpret_cast<unsigned short*>(out.data_ptr<at::BFloat16>()), B,
        C, H, W, (int)kh, (int)kw, (int)sh, (int)sw, (int)ph, (int)pw,
        (int)dh, (int)dw, oh, ow);
  } else if (x.scalar_type() == at::kBFloat16) {
    im2col_kernel<unsigned short><<<grid, block, 0, stream.stream()>>>(
        reinterpret_cast<const unsigned short*>(x.data_ptr<at::BFloat16>()),
        reinterpret_cast<unsigned short*>(out.data_ptr<at::BFloat16>()), B,
        C, H, W, (int)kh, (int)kw, (int)sh, (int)sw, (int)ph, (int)pw,
        (int)dh, (int)dw, oh, ow);
  } else {
    TORCH_CHECK(false, "im2col: dtype must be fp32 or bf16");
  }
  HIP_CHECK(hipGetLastError());
  return out;
}

// defined in jacobi_eigh.hip
int jacobi_eigh_max_dim();
std::vector<torch::Tensor> jacobi_eigh_batched(
    std::vector<torch::Tensor> mats);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X-native HIP kernels for distributed K-FAC";
  m.def("eigen_scale_", &eigen_scale_entry,
        "in-place V /= (dG dA^T + damping)");
  m.def("eigen_scale_batched_", &eigen_scale_batched_entry,
        "in-place batched V[b] /= (dG[b] dA[b]^T + damping)");
  m.def("syrk_factor_", &syrk_factor_entry,
        "fused bf16 MFMA factor product with bias column and running avg");
  m.def("im2col", &im2col_entry, "conv patch extraction to bf16 rows");
  m.def("jacobi_eigh_batched", &jacobi_eigh_batched,
        "batched LDS-resident Jacobi symmetric eigensolver (packed W, V)");
  m.def("jacobi_eigh_max_dim", &jacobi_eigh_max_dim,
        "largest dim the Jacobi kernel handles");
}
