// Batched blocked Householder tridiagonalization for K-FAC factors --
// the hand-written CDNA4 replacement for rocSOLVER's ssytrd on the
// hot path (the reference's eigensolve is cuSOLVER ssyevd,
// /root/reference/packages/tcmm/src/tcmm_kernel.cu:56-116; its
// tridiagonalization stage is ~80% of the solve and runs as ~3.7 us
// latency-bound latrd panel kernels -- ~18k launches per 4608 matrix).
//
// Design (validated numerically by scripts/sytrd_ref.py):
//
// * one PERSISTENT kernel launch per 64-column panel per bucket, batched
//   over all same-(padded-)dim factors: grid = (wgs_per_matrix, batch).
//   All 64 column iterations of the panel run INSIDE the launch with
//   per-matrix grid barriers (monotonic counter, relaxed sc1 poll +
//   s_sleep, agent-scope release/acquire fences) -- 2 barriers per
//   column instead of rocSOLVER's ~4 kernel launches per column; while
//   one matrix waits at its barrier the other matrices' workgroups
//   stream their matvecs, so the barrier latency hides in the batch.
// * full symmetric storage: the dominant read (the per-column trailing
//   matvec) streams CONTIGUOUS rows of the row-major torch tensor, one
//   wave per row, float4, perfectly coalesced. Row-major row i of the
//   symmetric input == column-major column i, so the output is exactly
//   LAPACK's uplo=LOWER sytrd format and rocsolver_sstedc plus the
//   torch-side WY back-transform consume it directly.
// * deferred-alpha panel algebra (scripts/sytrd_ref.py): W columns are
//   stored pre-alpha and every use folds the fix-up into per-column
//   scalar coefficients, which is what makes 2 barriers per column
//   sufficient (classical latrd needs 3 global syncs).
// * the V/W panel slabs for each workgroup's rows live in LDS (padded
//   stride, conflict-free for the lane-per-column reads); the rank-2*ib
//   trailing update is two rocblas_sgemm_strided_batched calls issued
//   by the host between panel launches (MFMA f32, GEMM-bound, a few %
//   of the time).
// * every spin is bounded: a barrier that exceeds its spin cap sets the
//   per-matrix status word, every other workgroup sees it and exits,
//   and the Python dispatch routes the bucket back to rocSOLVER -- the
//   kernel can go wrong slow, never hang the GPU.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#define SYTRD_NB 64          // panel width == max corrections per column
#define SYTRD_SLOTS 130      // scratch floats per workgroup:
                             // [0]=nrm2, [1..64]=pV, [65..128]=pW, [129]=pwv
#define SPIN_CAP 4000000     // ~0.5 s at ~300 cyc/poll: deadlock guard

namespace {

__device__ __forceinline__ float wave_reduce(float v) {
#pragma unroll
  for (int off = 32; off; off >>= 1) v += __shfl_down(v, off);
  return v;
}

// Per-matrix grid barrier: monotonic counter, target = phase * wgs.
// Returns false when the matrix has been aborted (status set).
__device__ bool grid_barrier(unsigned* cnt, int* status, int wgs,
                             unsigned phase) {
  // every storing wave drains its own outstanding global stores BEFORE
  // the leader's release fence (guide Appendix A hand-off checklist)
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
    // the compiler may drop the vmcnt wait behind buffer_wbl2 when its
    // scoreboard is provably empty; restate it (guide G16 pitfall 12)
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __hip_atomic_fetch_add(cnt, 1u, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
    const unsigned target = phase * (unsigned)wgs;
    long spins = 0;
    while (__hip_atomic_load(cnt, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT) < target) {
      // escalating backoff: hot polls only for the first ~2 us, then
      // ~1.7 us sleeps -- hundreds of concurrent pollers on a few
      // counter words would otherwise steal chip bandwidth from the
      // matvec streams (guide: polling-cost row)
      if (spins < 32)
        __builtin_amdgcn_s_sleep(4);
      else
        __builtin_amdgcn_s_sleep(64);
      if ((++spins & 255) == 0) {
        if (__hip_atomic_load(status, __ATOMIC_RELAXED,
                              __HIP_MEMORY_SCOPE_AGENT) != 0) break;
        if (spins > SPIN_CAP) {
          __hip_atomic_store(status, 1, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT);
          break;
        }
      }
    }
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  }
  __syncthreads();
  // abort propagates: everyone re-checks status after the fence
  __shared__ int s_abort[1];
  if (threadIdx.x == 0)
    s_abort[0] = __hip_atomic_load(status, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
  __syncthreads();
  return s_abort[0] == 0;
}

// One panel of the blocked tridiagonalization (columns j0..j0+ib-1) for
// every matrix in the bucket.  blockIdx.x = workgroup within matrix,
// blockIdx.y = matrix.  Each workgroup owns rows [r0, r1) of its
// matrix; RS = padded LDS slab stride (odd).
__global__ __launch_bounds__(256) void latrd_panel_kernel(
    float* __restrict__ A, float* __restrict__ W,
    float* __restrict__ E, float* __restrict__ tau,
    float* __restrict__ scratch, unsigned* __restrict__ syncc,
    int* __restrict__ status, int n, int j0, int ib, int R,
    int wgs_alloc, int mode) {
  const int bm = blockIdx.y;
  const int w = blockIdx.x;
  const int wgs = gridDim.x;
  float* A_m = A + (size_t)bm * n * n;
  float* W_m = W + (size_t)bm * SYTRD_NB * n;
  float* scr = scratch + (size_t)bm * wgs_alloc * SYTRD_SLOTS;
  unsigned* cnt = syncc + bm;
  int* st = status + bm;
  const int r0 = j0 + w * R;
  const int r1 = min(r0 + R, n);
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  extern __shared__ __attribute__((aligned(16))) float smem[];
  const int RS = R | 1;  // odd stride -> conflict-free [c][r] lane reads
  float* slabV = smem;                       // [SYTRD_NB][RS]
  float* slabW = slabV + SYTRD_NB * RS;      // [SYTRD_NB][RS]
  float* x_own = slabW + SYTRD_NB * RS;      // [RS]
  float* sums = x_own + RS;                  // [SYTRD_SLOTS]
  float* coefA = sums + SYTRD_SLOTS;         // [SYTRD_NB]
  float* coefB = coefA + SYTRD_NB;           // [SYTRD_NB]
  float* qA = coefB + SYTRD_NB;              // [SYTRD_NB]
  float* qB = qA + SYTRD_NB;                 // [SYTRD_NB]
  float* alpha = qB + SYTRD_NB;              // [SYTRD_NB]
  float* red = alpha + SYTRD_NB;             // [4] cross-wave scratch
  // the pre-scale x row, staged once per column (16B-aligned carve)
  float* x_lds = (float*)((((size_t)(red + 4)) + 15) & ~(size_t)15);

  if (__hip_atomic_load(st, __ATOMIC_RELAXED,
                        __HIP_MEMORY_SCOPE_AGENT) != 0)
    return;  // a previous panel aborted this matrix

  float tau_prev = 0.f, s_prev = 0.f;
  unsigned phase = 0;

  for (int i = 0; i < ib; ++i) {
    const int j = j0 + i;
    if (mode & 2) {  // ablation: barriers only
      if (!grid_barrier(cnt, st, wgs, ++phase)) return;
      if (!grid_barrier(cnt, st, wgs, ++phase)) return;
      continue;
    }
    // ================= phase A =================
    if (i > 0) {
      // alpha of the previous column from the pwv partials
      float p = 0.f;
      for (int t = tid; t < wgs; t += 256)
        p += scr[(size_t)129 * wgs_alloc + t];
      p = wave_reduce(p);
      if (lane == 0) red[wave] = p;
      __syncthreads();
      if (tid == 0)
        alpha[i - 1] = -0.5f * tau_prev * (red[0] + red[1] +
                                           red[2] + red[3]);
      __syncthreads();
      // scaled-v writeback of column j-1 + slab append (own rows)
      float* Aprev = A_m + (size_t)(j - 1) * n;
      float* Wprev = W_m + (size_t)(i - 1) * n;
      for (int r = max(r0, j) + tid; r < r1; r += 256) {
        float v = (r == j) ? 1.f : s_prev * Aprev[r];
        Aprev[r] = v;
        slabV[(i - 1) * RS + (r - r0)] = v;
        slabW[(i - 1) * RS + (r - r0)] = Wprev[r];
      }
      // per-column correction coefficients (one wave, lane = c).
      // c == i-1's V value at row j IS the unit element (j = (j-1)+1)
      // and its stored copy is being written by another workgroup in
      // THIS phase -- use the exact 1.0, never the racy load.
      if (wave == 0 && lane < i) {
        float cb = (lane == i - 1)
                       ? 1.f
                       : A_m[(size_t)(j0 + lane) * n + j];
        coefA[lane] = W_m[(size_t)lane * n + j] + 2.f * alpha[lane] * cb;
        coefB[lane] = cb;
      }
    }
    __syncthreads();
    // x-correction: waves over rows, lanes over c (shfl-reduced)
    {
      float* Aj = A_m + (size_t)j * n;
      const int rlo = max(r0, j);
      for (int r = rlo + wave; r < r1; r += 4) {
        const int rl = r - r0;
        float corr = 0.f;
        if (lane < i)
          corr = slabV[lane * RS + rl] * coefA[lane] +
                 slabW[lane * RS + rl] * coefB[lane];
        corr = wave_reduce(corr);
        if (lane == 0) {
          float x = Aj[r] - corr;
          Aj[r] = x;
          x_own[rl] = x;
        }
      }
    }
    __syncthreads();
    // partials: nrm2 (r >= j+2), pV/pW (r >= j+1), written to scratch
    {
      float nrm = 0.f;
      for (int r = max(r0, j + 2) + tid; r < r1; r += 256) {
        float x = x_own[r - r0];
        nrm += x * x;
      }
      nrm = wave_reduce(nrm);
      if (lane == 0) red[wave] = nrm;
      __syncthreads();
      if (tid == 0) {
        float t0 = red[0] + red[1] + red[2] + red[3];
        if (!isfinite(t0))
          __hip_atomic_store(st, 2, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT);
        scr[(size_t)0 * wgs_alloc + w] = t0;
      }
      // pV/pW: waves split the c-range, lanes... lanes = c, loop rows.
      const int rlo = max(r0, j + 1);
      for (int c = wave; c < i; c += 4) {
        float pv = 0.f, pw = 0.f;
        for (int rl = rlo - r0 + lane; rl < r1 - r0; rl += 64) {
          float x = x_own[rl];
          pv += slabV[c * RS + rl] * x;
          pw += slabW[c * RS + rl] * x;
        }
        pv = wave_reduce(pv);
        pw = wave_reduce(pw);
        if (lane == 0) {
          scr[(size_t)(1 + c) * wgs_alloc + w] = pv;
          scr[(size_t)(65 + c) * wgs_alloc + w] = pw;
        }
      }
    }
    if (!grid_barrier(cnt, st, wgs, ++phase)) return;
    // ================= phase B =================
    // sum partials across workgroups (only live slots)
    {
      if (tid < SYTRD_SLOTS - 1) {
        const int s = tid;
        bool live = (s == 0) || (s >= 1 && s <= i) ||
                    (s >= 65 && s <= 64 + i);
        if (live) {
          float acc = 0.f;
          for (int t = 0; t < wgs; ++t)
            acc += scr[(size_t)s * wgs_alloc + t];
          sums[s] = acc;
        }
      }
      __syncthreads();
    }
    // larfg scalars, redundant per thread
    const float x1 = A_m[(size_t)j * n + j + 1];
    const float nrm2 = sums[0];
    float beta, tau_j, s;
    if (nrm2 == 0.f) {
      beta = x1; tau_j = 0.f; s = 0.f;
    } else {
      beta = -copysignf(sqrtf(x1 * x1 + nrm2), x1 == 0.f ? 1.f : x1);
      tau_j = (beta - x1) / beta;
      s = 1.f / (x1 - beta);
    }
    if (w == 0 && tid == 0) {
      if (!isfinite(beta))
        __hip_atomic_store(st, 2, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
      E[(size_t)bm * n + j] = beta;
      tau[(size_t)bm * n + j] = tau_j;
    }
    // matvec correction coefficients (one wave, lane = c)
    if (wave == 0 && lane < i) {
      float vj1 = A_m[(size_t)(j0 + lane) * n + j + 1];
      float wj1 = W_m[(size_t)lane * n + j + 1];
      float sV = s * (sums[1 + lane] - vj1 * x1) + vj1;
      float sW = s * (sums[65 + lane] - wj1 * x1) + wj1;
      qA[lane] = sW + 2.f * alpha[lane] * sV;
      qB[lane] = sV;
    }
    // stage the pre-scale x row into LDS once per column: the matvec
    // would otherwise re-read it from L2 per row, doubling the VMEM
    // issue in the hot loop (per-CU streaming is issue-bound)
    {
      const float* Aj = A_m + (size_t)j * n;
      for (int t4 = j0 + 4 * tid; t4 < n; t4 += 1024)
        *(float4*)(x_lds + (t4 - j0)) = *(const float4*)(Aj + t4);
    }
    __syncthreads();
    // the trailing matvec: each wave streams FOUR rows concurrently
    // (independent load/acc chains -- a single row's loads serialize
    // behind its reduction and pay full HBM latency per row), x from
    // LDS shared across the group
    {
      const int rlo = max(r0, j + 1);
      float pwv_acc = 0.f;
      for (int rg = rlo + 4 * wave; rg < r1; rg += 16) {
        const int nr = min(4, r1 - rg);
        // rows rg..rg+nr-1; k >= nr alias row rg (loads valid,
        // results discarded)
        const float* Ar[4];
#pragma unroll
        for (int k = 0; k < 4; ++k)
          Ar[k] = A_m + (size_t)(k < nr ? rg + k : rg) * n;
        float dot[4] = {0.f, 0.f, 0.f, 0.f};
        int t = j + 2;
        // head to 16B alignment (n % 4 == 0 -> row bases aligned)
        if (lane == 0)
          for (; t < n && (t & 3); ++t) {
            const float xh = x_lds[t - j0];
#pragma unroll
            for (int k = 0; k < 4; ++k) dot[k] += Ar[k][t] * xh;
          }
        t = (j + 2 + 3) & ~3;
        if ((mode & 1) == 0) {
          float4 acc[4];
#pragma unroll
          for (int k = 0; k < 4; ++k)
            acc[k] = {0.f, 0.f, 0.f, 0.f};
          for (int tb = t + 4 * lane; tb < n; tb += 256) {
            const float4 xv = *(const float4*)(x_lds + (tb - j0));
#pragma unroll
            for (int k = 0; k < 4; ++k) {
              const float4 a = *(const float4*)(Ar[k] + tb);
              acc[k].x = fmaf(a.x, xv.x, acc[k].x);
              acc[k].y = fmaf(a.y, xv.y, acc[k].y);
              acc[k].z = fmaf(a.z, xv.z, acc[k].z);
              acc[k].w = fmaf(a.w, xv.w, acc[k].w);
            }
          }
#pragma unroll
          for (int k = 0; k < 4; ++k)
            dot[k] += (acc[k].x + acc[k].y) + (acc[k].z + acc[k].w);
        }
#pragma unroll
        for (int k = 0; k < 4; ++k) {
          if (k >= nr) break;
          const int r = rg + k;
          // fold: w_pre = A[r][j+1] + s*dot - corrections (lane = c)
          float contrib = s * dot[k];
          const int rl = r - r0;
          if (lane < i)
            contrib -= slabV[lane * RS + rl] * qA[lane] +
                       slabW[lane * RS + rl] * qB[lane];
          if (lane == 0) contrib += Ar[k][j + 1];
          contrib = wave_reduce(contrib);
          if (lane == 0) {
            float w1 = tau_j * contrib;
            W_m[(size_t)i * n + r] = w1;
            float vt = (r == j + 1) ? 1.f : s * x_own[rl];
            pwv_acc += w1 * vt;
          }
        }
      }
      pwv_acc = wave_reduce(pwv_acc);  // lane0-only values; harmless
      if (lane == 0) red[wave] = pwv_acc;
      __syncthreads();
      if (tid == 0)
        scr[(size_t)129 * wgs_alloc + w] =
            red[0] + red[1] + red[2] + red[3];
    }
    tau_prev = tau_j;
    s_prev = s;
    if (!grid_barrier(cnt, st, wgs, ++phase)) return;
  }
  // ================= finalize =================
  // alpha of the last column, scaled writeback of its v, W += alpha*v
  {
    float p = 0.f;
    for (int t = tid; t < wgs; t += 256)
      p += scr[(size_t)129 * wgs_alloc + t];
    p = wave_reduce(p);
    if (lane == 0) red[wave] = p;
    __syncthreads();
    if (tid == 0)
      alpha[ib - 1] = -0.5f * tau_prev * (red[0] + red[1] +
                                          red[2] + red[3]);
    __syncthreads();
    const int jl = j0 + ib - 1;
    float* Al = A_m + (size_t)jl * n;
    for (int r = max(r0, jl + 1) + tid; r < r1; r += 256) {
      float v = (r == jl + 1) ? 1.f : s_prev * Al[r];
      Al[r] = v;
    }
    __syncthreads();
    for (int c = 0; c < ib; ++c) {
      const float a = alpha[c];
      const int jc = j0 + c;
      float* Wc = W_m + (size_t)c * n;
      if (c == ib - 1) {
        for (int r = max(r0, jc + 1) + tid; r < r1; r += 256)
          Wc[r] += a * Al[r];
      } else {
        for (int r = max(r0, jc + 1) + tid; r < r1; r += 256)
          Wc[r] += a * slabV[c * RS + (r - r0)];
      }
    }
  }
}

}  // namespace

// Host driver: full blocked tridiagonalization of As (b, n, n) fp32
// contiguous row-major symmetric, n % 4 == 0.  On exit As holds the
// scaled reflectors (row j: unit at j+1, v at j+2.., LAPACK lower
// format when read column-major) with D on the diagonal; returns
// (E (b,n), tau (b,n), status (b,) int32 -- nonzero = aborted, caller
// must fall back).  Runs on the torch current stream; the trailing
// rank-2*ib updates are issued by the caller (Python) between panels
// via sgemm -- no, by THIS driver via rocblas (see rocsolver_eig.hip
// for the shared handle).
std::vector<torch::Tensor> sytrd_panels_device(
    torch::Tensor As, torch::Tensor W, torch::Tensor E,
    torch::Tensor tau, torch::Tensor scratch, torch::Tensor syncc,
    torch::Tensor status, int j0, int ib, int R, int wgs_alloc,
    int wgs, int mode) {
  const int n = (int)As.size(1);
  const int b = (int)As.size(0);
  auto stream = c10::hip::getCurrentHIPStream();
  const int RS = R | 1;
  const size_t lds = sizeof(float) *
      (2 * SYTRD_NB * RS + RS + SYTRD_SLOTS + 5 * SYTRD_NB + 4 +
       (size_t)(n - j0)) + 16;
  hipError_t err = hipMemsetAsync(syncc.data_ptr(), 0,
                                  sizeof(unsigned) * b, stream.stream());
  TORCH_CHECK(err == hipSuccess, "sytrd memset: ",
              hipGetErrorString(err));
  latrd_panel_kernel<<<dim3(wgs, b), 256, lds, stream.stream()>>>(
      As.data_ptr<float>(), W.data_ptr<float>(), E.data_ptr<float>(),
      tau.data_ptr<float>(), scratch.data_ptr<float>(),
      (unsigned*)syncc.data_ptr(), status.data_ptr<int>(), n, j0, ib, R,
      wgs_alloc, mode);
  err = hipGetLastError();
  TORCH_CHECK(err == hipSuccess, "latrd_panel_kernel launch: ",
              hipGetErrorString(err));
  return {};
}

int sytrd_panel_max_blocks(int R, int xlen_max) {
  const int RS = R | 1;
  const size_t lds = sizeof(float) *
      (2 * SYTRD_NB * RS + RS + SYTRD_SLOTS + 5 * SYTRD_NB + 4 +
       (size_t)xlen_max) + 16;
  int nb = 0;
  hipError_t err = hipOccupancyMaxActiveBlocksPerMultiprocessor(
      &nb, (const void*)latrd_panel_kernel, 256, lds);
  if (err != hipSuccess) return 0;
  int cus = 0;
  hipDeviceProp_t prop;
  if (hipGetDeviceProperties(&prop, 0) == hipSuccess)
    cus = prop.multiProcessorCount;
  if (cus <= 0) cus = 256;
  // the occupancy API over-reports by one block/CU for SGPR-heavy
  // kernels on ROCm 7.2 (guide: residency & cooperative launch); a
  // non-resident workgroup deadlocks the grid barrier, so keep a
  // one-block margin whenever we can afford it
  if (nb > 1) nb -= 1;
  return nb * cus;
}
