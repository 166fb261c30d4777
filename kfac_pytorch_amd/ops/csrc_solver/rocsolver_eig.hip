// Async / batched symmetric eigensolves for K-FAC factors via rocSOLVER,
// driven the MI355X way: device-resident arguments, persistent
// handle+stream pool, NO host synchronization inside the hot path.
//
// Why this exists: torch.linalg.eigh's rocSOLVER wrapper host-syncs
// every call (info check + workspace staging), so a ResNet-50 rank's
// ~106 per-layer eigensolves serialize into ~1.7 s/step.  rocSOLVER
// itself is stream-ordered with all-GPU arguments; issuing each solve
// on a pool stream and joining once overlaps them, and factor dims
// repeat heavily across a network, so same-dim factors additionally
// batch into ONE ``syevdj_strided_batched`` call.
//
// This is the library tier of the eigensolver stack (the hand-written
// LDS-Jacobi kernel in csrc/jacobi_eigh.hip covers m <= 128); it
// replaces the reference's per-layer cuSOLVER ``cusolverDnSsyevd``
// (reference: packages/tcmm/src/tcmm_kernel.cu:56-116 -- which even
// cudaMalloc's its workspace and device-syncs per call).
//
// Layout note: rocSOLVER is column-major.  Inputs are symmetric, so no
// transpose is needed going in; outputs leave eigenvectors in the
// matrix buffer column-major, i.e. interpreting the same buffer
// row-major, ROW i is the i-th eigenvector.  The Python wrapper
// returns ``V.mT`` to restore the eigh contract (columns =
// eigenvectors) without a copy.

#include <hip/hip_runtime.h>
#include <rocblas/rocblas.h>
#include <rocsolver/rocsolver.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <c10/hip/HIPCachingAllocator.h>

#include <algorithm>
#include <cstdlib>
#include <sstream>
#include <vector>

// Tell the caching allocator a tensor is in use on one of our pool
// streams, so freeing it on the torch stream cannot hand its memory to
// another op while the async rocSOLVER call is still running.
static void record_on(const at::Tensor& t, hipStream_t s) {
  c10::hip::HIPCachingAllocator::recordStream(
      t.storage().data_ptr(),
      c10::hip::getStreamFromExternal(s, t.get_device()));
}

#define HIPCHECK(cmd)                                                     \
  do {                                                                    \
    hipError_t e_ = (cmd);                                                \
    TORCH_CHECK(e_ == hipSuccess, "HIP error ", hipGetErrorString(e_),    \
                " at " __FILE__ ":", __LINE__);                           \
  } while (0)

#define ROCBLASCHECK(cmd)                                                 \
  do {                                                                    \
    rocblas_status s_ = (cmd);                                            \
    TORCH_CHECK(s_ == rocblas_status_success, "rocSOLVER error ", (int)s_,\
                " at " __FILE__ ":", __LINE__);                           \
  } while (0)

namespace {

struct Slot {
  rocblas_handle handle = nullptr;
  hipStream_t stream = nullptr;
  hipEvent_t event = nullptr;
  void* workspace = nullptr;
};

constexpr int POOL = 8;
// Persistent per-handle device workspace.  Without it rocBLAS
// auto-allocates (hipMalloc/hipFree) inside every rocSOLVER call --
// device-synchronizing operations that serialize the whole pool and
// kill cross-stream overlap.  4 GB per handle (36 GB total of the
// 288 GB HBM) covers batched ssyevd groups at m ~ 4.6k; a
// request beyond it surfaces as rocblas_status_memory_error, which the
// Python dispatch catches and reroutes to the syevd pool.
constexpr size_t WORKSPACE_BYTES = size_t(4) << 30;

Slot g_pool[POOL];
rocblas_handle g_main_handle = nullptr;  // bound to torch current stream
void* g_main_workspace = nullptr;
hipEvent_t g_acq_event = nullptr;
bool g_init = false;

void ensure_init() {
  if (g_init) return;
  for (int i = 0; i < POOL; ++i) {
    ROCBLASCHECK(rocblas_create_handle(&g_pool[i].handle));
    HIPCHECK(hipStreamCreateWithFlags(&g_pool[i].stream,
                                      hipStreamNonBlocking));
    HIPCHECK(hipEventCreateWithFlags(&g_pool[i].event,
                                     hipEventDisableTiming));
    ROCBLASCHECK(rocblas_set_stream(g_pool[i].handle, g_pool[i].stream));
    HIPCHECK(hipMalloc(&g_pool[i].workspace, WORKSPACE_BYTES));
    ROCBLASCHECK(rocblas_set_workspace(g_pool[i].handle,
                                       g_pool[i].workspace,
                                       WORKSPACE_BYTES));
  }
  ROCBLASCHECK(rocblas_create_handle(&g_main_handle));
  HIPCHECK(hipMalloc(&g_main_workspace, WORKSPACE_BYTES));
  ROCBLASCHECK(rocblas_set_workspace(g_main_handle, g_main_workspace,
                                     WORKSPACE_BYTES));
  HIPCHECK(hipEventCreateWithFlags(&g_acq_event, hipEventDisableTiming));
  g_init = true;
}

}  // namespace

// Batched eigensolve of same-size matrices: As (b, n, n) fp32 contiguous,
// OVERWRITTEN with eigenvectors (column-major -> row i = eigenvector i
// when read row-major).  Returns W (b, n) ascending and the info tensor
// (b,) for deferred validation.  ``slot`` >= 0 issues the call on that
// pool stream (event-ordered after the torch stream; caller must
// ``join_pool_`` before consuming) so multiple dim-groups and the
// mixed-size singles overlap; slot < 0 runs on the torch current stream.
std::vector<torch::Tensor> syevdj_batched_(torch::Tensor As, int slot) {
  ensure_init();
  TORCH_CHECK(As.is_cuda() && As.dim() == 3 && As.size(1) == As.size(2),
              "syevdj_batched_: (b, n, n) GPU tensor required");
  TORCH_CHECK(As.scalar_type() == at::kFloat && As.is_contiguous(),
              "syevdj_batched_: fp32 contiguous required");
  const long b = As.size(0);
  const long n = As.size(1);
  auto W = at::empty({b, n}, As.options());
  auto info = at::empty({b}, As.options().dtype(at::kInt));
  auto stream = c10::hip::getCurrentHIPStream();
  if (slot < 0) {
    ROCBLASCHECK(rocblas_set_stream(g_main_handle, stream.stream()));
    ROCBLASCHECK(rocsolver_ssyevdj_strided_batched(
        g_main_handle, rocblas_evect_original, rocblas_fill_lower,
        (rocblas_int)n, As.data_ptr<float>(), (rocblas_int)n,
        (rocblas_stride)(n * n), W.data_ptr<float>(), (rocblas_stride)n,
        info.data_ptr<int>(), (rocblas_int)b));
    return {W, info};
  }
  Slot& s = g_pool[slot % POOL];
  HIPCHECK(hipEventRecord(g_acq_event, stream.stream()));
  HIPCHECK(hipStreamWaitEvent(s.stream, g_acq_event, 0));
  record_on(As, s.stream);
  record_on(W, s.stream);
  record_on(info, s.stream);
  ROCBLASCHECK(rocsolver_ssyevdj_strided_batched(
      s.handle, rocblas_evect_original, rocblas_fill_lower,
      (rocblas_int)n, As.data_ptr<float>(), (rocblas_int)n,
      (rocblas_stride)(n * n), W.data_ptr<float>(), (rocblas_stride)n,
      info.data_ptr<int>(), (rocblas_int)b));
  return {W, info};
}

// Batched divide-and-conquer eigensolve of same-size matrices (the
// syevd algorithm, batched: one call runs every matrix's tiny
// latency-bound tridiagonalization panels concurrently -- rocprof
// shows the single-matrix path spends its time in ~3.7 us latrd
// kernels).  Same layout contract as syevdj_batched_.
std::vector<torch::Tensor> syevd_batched_(torch::Tensor As, int slot) {
  ensure_init();
  TORCH_CHECK(As.is_cuda() && As.dim() == 3 && As.size(1) == As.size(2),
              "syevd_batched_: (b, n, n) GPU tensor required");
  TORCH_CHECK(As.scalar_type() == at::kFloat && As.is_contiguous(),
              "syevd_batched_: fp32 contiguous required");
  const long b = As.size(0);
  const long n = As.size(1);
  auto W = at::empty({b, n}, As.options());
  auto E = at::empty({b, n}, As.options());
  auto info = at::empty({b}, As.options().dtype(at::kInt));
  auto stream = c10::hip::getCurrentHIPStream();
  rocblas_handle h = g_main_handle;
  if (slot >= 0) {
    Slot& s = g_pool[slot % POOL];
    HIPCHECK(hipEventRecord(g_acq_event, stream.stream()));
    HIPCHECK(hipStreamWaitEvent(s.stream, g_acq_event, 0));
    record_on(As, s.stream);
    record_on(W, s.stream);
    record_on(E, s.stream);
    record_on(info, s.stream);
    h = s.handle;
  } else {
    ROCBLASCHECK(rocblas_set_stream(g_main_handle, stream.stream()));
  }
  ROCBLASCHECK(rocsolver_ssyevd_strided_batched(
      h, rocblas_evect_original, rocblas_fill_lower, (rocblas_int)n,
      As.data_ptr<float>(), (rocblas_int)n, (rocblas_stride)(n * n),
      W.data_ptr<float>(), (rocblas_stride)n, E.data_ptr<float>(),
      (rocblas_stride)n, info.data_ptr<int>(), (rocblas_int)b));
  return {W, info};
}

// Make the torch current stream wait on every pool stream (call after a
// burst of slot-issued syevdj/syevd work).
void join_pool_() {
  ensure_init();
  auto stream = c10::hip::getCurrentHIPStream();
  for (int i = 0; i < POOL; ++i) {
    HIPCHECK(hipEventRecord(g_pool[i].event, g_pool[i].stream));
    HIPCHECK(hipStreamWaitEvent(stream.stream(), g_pool[i].event, 0));
  }
}

// Pool-parallel eigensolves of differently-sized matrices: each matrix
// is issued on one of POOL internal streams (round-robin) so the
// host-async rocSOLVER calls overlap on-device; ``join`` makes the torch
// current stream wait on every pool stream.  mats are OVERWRITTEN with
// eigenvectors (same layout note as above).  Returns per-matrix W plus
// one packed info tensor.
std::vector<torch::Tensor> syevd_pool_(std::vector<torch::Tensor> mats) {
  ensure_init();
  TORCH_CHECK(!mats.empty(), "syevd_pool_: empty batch");
  auto stream = c10::hip::getCurrentHIPStream();
  // order pool streams after pending torch-stream work (factor updates)
  HIPCHECK(hipEventRecord(g_acq_event, stream.stream()));
  for (int i = 0; i < POOL; ++i) {
    HIPCHECK(hipStreamWaitEvent(g_pool[i].stream, g_acq_event, 0));
  }
  auto opts = mats[0].options();
  auto info = at::empty({(long)mats.size()}, opts.dtype(at::kInt));
  std::vector<torch::Tensor> out;
  out.reserve(mats.size() + 1);
  std::vector<torch::Tensor> workE;
  for (size_t i = 0; i < mats.size(); ++i) {
    auto& A = mats[i];
    TORCH_CHECK(A.is_cuda() && A.dim() == 2 && A.size(0) == A.size(1),
                "syevd_pool_: square GPU matrices required");
    TORCH_CHECK(A.scalar_type() == at::kFloat && A.is_contiguous(),
                "syevd_pool_: fp32 contiguous required");
    const long n = A.size(0);
    auto W = at::empty({n}, opts);
    auto E = at::empty({n}, opts);
    workE.push_back(E);
    Slot& s = g_pool[i % POOL];
    record_on(A, s.stream);
    record_on(W, s.stream);
    record_on(E, s.stream);
    record_on(info, s.stream);
    ROCBLASCHECK(rocsolver_ssyevd(
        s.handle, rocblas_evect_original, rocblas_fill_lower,
        (rocblas_int)n, A.data_ptr<float>(), (rocblas_int)n,
        W.data_ptr<float>(), E.data_ptr<float>(),
        info.data_ptr<int>() + i));
    out.push_back(W);
  }
  // join: torch stream waits on all pool streams (all of them -- the
  // caller may have issued group work on slots this call didn't use)
  for (int i = 0; i < POOL; ++i) {
    HIPCHECK(hipEventRecord(g_pool[i].event, g_pool[i].stream));
    HIPCHECK(hipStreamWaitEvent(stream.stream(), g_pool[i].event, 0));
  }
  out.push_back(info);
  return out;
}

// Pool-parallel Cholesky inverse (potrf + potri) for the 'inverse'
// K-FAC family: damped SPD factors, in-place, overlapped on the pool
// (replaces the serial per-layer torch.cholesky_inverse loop;
// reference: kfac/utils.py:11-20, kfac_preconditioner_inv.py:109-129).
std::vector<torch::Tensor> potri_pool_(std::vector<torch::Tensor> mats) {
  ensure_init();
  TORCH_CHECK(!mats.empty(), "potri_pool_: empty batch");
  auto stream = c10::hip::getCurrentHIPStream();
  HIPCHECK(hipEventRecord(g_acq_event, stream.stream()));
  for (int i = 0; i < POOL; ++i) {
    HIPCHECK(hipStreamWaitEvent(g_pool[i].stream, g_acq_event, 0));
  }
  auto info = at::empty({(long)mats.size() * 2},
                        mats[0].options().dtype(at::kInt));
  for (size_t i = 0; i < mats.size(); ++i) {
    auto& A = mats[i];
    TORCH_CHECK(A.is_cuda() && A.dim() == 2 && A.size(0) == A.size(1) &&
                    A.scalar_type() == at::kFloat && A.is_contiguous(),
                "potri_pool_: square fp32 contiguous GPU matrices required");
    const long n = A.size(0);
    Slot& s = g_pool[i % POOL];
    record_on(A, s.stream);
    record_on(info, s.stream);
    ROCBLASCHECK(rocsolver_spotrf(s.handle, rocblas_fill_lower,
                                  (rocblas_int)n, A.data_ptr<float>(),
                                  (rocblas_int)n,
                                  info.data_ptr<int>() + 2 * i));
    ROCBLASCHECK(rocsolver_spotri(s.handle, rocblas_fill_lower,
                                  (rocblas_int)n, A.data_ptr<float>(),
                                  (rocblas_int)n,
                                  info.data_ptr<int>() + 2 * i + 1));
  }
  // join ALL pool streams (not just the ones this call used): callers
  // may have issued batched work on other slots before calling in, and
  // they rely on "potri_pool_ joins the whole pool" before mirroring
  // results on the torch stream.
  for (int i = 0; i < POOL; ++i) {
    HIPCHECK(hipEventRecord(g_pool[i].event, g_pool[i].stream));
    HIPCHECK(hipStreamWaitEvent(stream.stream(), g_pool[i].event, 0));
  }
  return {info};
}

// Batched Cholesky inverse of same-size SPD matrices (potrf + potri,
// strided-batched): As (b, n, n) fp32 contiguous, OVERWRITTEN with the
// inverse's column-major LOWER triangle = row-major UPPER triangle
// (mirror in the caller).  slot semantics as syevd_batched_.
std::vector<torch::Tensor> potri_batched_(torch::Tensor As, int slot) {
  ensure_init();
  TORCH_CHECK(As.is_cuda() && As.dim() == 3 && As.size(1) == As.size(2),
              "potri_batched_: (b, n, n) GPU tensor required");
  TORCH_CHECK(As.scalar_type() == at::kFloat && As.is_contiguous(),
              "potri_batched_: fp32 contiguous required");
  const long b = As.size(0);
  const long n = As.size(1);
  auto info = at::empty({2 * b}, As.options().dtype(at::kInt));
  auto stream = c10::hip::getCurrentHIPStream();
  rocblas_handle h = g_main_handle;
  if (slot >= 0) {
    Slot& s = g_pool[slot % POOL];
    HIPCHECK(hipEventRecord(g_acq_event, stream.stream()));
    HIPCHECK(hipStreamWaitEvent(s.stream, g_acq_event, 0));
    record_on(As, s.stream);
    record_on(info, s.stream);
    h = s.handle;
  } else {
    ROCBLASCHECK(rocblas_set_stream(g_main_handle, stream.stream()));
  }
  ROCBLASCHECK(rocsolver_spotrf_strided_batched(
      h, rocblas_fill_lower, (rocblas_int)n, As.data_ptr<float>(),
      (rocblas_int)n, (rocblas_stride)(n * n), info.data_ptr<int>(),
      (rocblas_int)b));
  ROCBLASCHECK(rocsolver_spotri_strided_batched(
      h, rocblas_fill_lower, (rocblas_int)n, As.data_ptr<float>(),
      (rocblas_int)n, (rocblas_stride)(n * n),
      info.data_ptr<int>() + b, (rocblas_int)b));
  return {info};
}

// The three stages of a symmetric eigensolve, exposed separately so
// the Python side can (a) profile the sytrd/stedc/ormtr split and
// (b) swap any stage for a custom kernel (round-2 SBR plan: replace
// sytrd, keep library D&C + back-transform).  All run on the torch
// current stream via the main handle.
std::vector<torch::Tensor> sytrd_(torch::Tensor A) {
  ensure_init();
  TORCH_CHECK(A.is_cuda() && A.dim() == 2 && A.size(0) == A.size(1) &&
                  A.scalar_type() == at::kFloat && A.is_contiguous(),
              "sytrd_: square fp32 contiguous GPU matrix required");
  const long n = A.size(0);
  auto D = at::empty({n}, A.options());
  auto E = at::empty({n}, A.options());
  auto tau = at::empty({n}, A.options());
  auto stream = c10::hip::getCurrentHIPStream();
  ROCBLASCHECK(rocblas_set_stream(g_main_handle, stream.stream()));
  ROCBLASCHECK(rocsolver_ssytrd(g_main_handle, rocblas_fill_lower,
                                (rocblas_int)n, A.data_ptr<float>(),
                                (rocblas_int)n, D.data_ptr<float>(),
                                E.data_ptr<float>(),
                                tau.data_ptr<float>()));
  return {D, E, tau};
}

// defined in sytrd_panel.hip
std::vector<torch::Tensor> sytrd_panels_device(
    torch::Tensor As, torch::Tensor W, torch::Tensor E,
    torch::Tensor tau, torch::Tensor scratch, torch::Tensor syncc,
    torch::Tensor status, int j0, int ib, int R, int wgs_alloc, int wgs,
    int mode);
int sytrd_panel_max_blocks(int R, int xlen_max);

// Custom batched blocked tridiagonalization (sytrd_panel.hip): the
// hand-written persistent-panel kernel replaces rocSOLVER's latrd
// launch storm; the rank-2*ib trailing updates run as two
// sgemm_strided_batched per panel on the same (torch) stream.  As
// (b, n, n) fp32 contiguous symmetric row-major, n % 4 == 0, is
// overwritten with scaled reflectors (LAPACK lower format when read
// column-major; D stays on the diagonal).  Returns (E, tau, status);
// a nonzero status entry means that matrix aborted (bounded-spin
// barrier or non-finite data) and the caller must fall back.
std::vector<torch::Tensor> sytrd_batched_custom_(torch::Tensor As) {
  ensure_init();
  TORCH_CHECK(As.is_cuda() && As.dim() == 3 && As.size(1) == As.size(2),
              "sytrd_batched_custom_: (b, n, n) GPU tensor required");
  TORCH_CHECK(As.scalar_type() == at::kFloat && As.is_contiguous(),
              "sytrd_batched_custom_: fp32 contiguous required");
  const int b = (int)As.size(0);
  const int n = (int)As.size(1);
  TORCH_CHECK(n % 4 == 0 && n >= 128,
              "sytrd_batched_custom_: n % 4 == 0 and n >= 128 required");
  // pick rows-per-workgroup: aim for ~512 TOTAL workgroups across the
  // batch (fills the chip's bandwidth) but no more -- the per-column
  // cross-workgroup reductions cost O(wgs^2) scratch reads per matrix
  // and every extra workgroup is another barrier poller, both measured
  // dominant when wgs_per_matrix was maximized.  The grid must also be
  // guaranteed co-resident (a non-resident workgroup would deadlock
  // the grid barrier).
  static const int cand[] = {24, 32, 36, 48, 64, 96, 128};
  int total_target = 384;
  if (const char* e = getenv("KFAC_SYTRD_WGS"))
    total_target = std::max(64, atoi(e));
  const int want = std::min(192, std::max(1, total_target / b));
  int R = 0, wgs = 0;
  for (int c : cand) {
    int w = (n + c - 1) / c;
    if (w > want) continue;
    int cp = sytrd_panel_max_blocks(c, n);
    if ((long)w * b <= cp) { R = c; wgs = w; break; }
  }
  if (R == 0) {
    // batch too large for the target: take the coarsest grid that fits
    int w = (n + 127) / 128;
    int cp = sytrd_panel_max_blocks(128, n);
    if ((long)w * b <= cp) { R = 128; wgs = w; }
  }
  TORCH_CHECK(R > 0, "sytrd_batched_custom_: no resident grid for n=",
              n, " b=", b, " (cap(128)=", sytrd_panel_max_blocks(128, n),
              ") -- fall back to rocSOLVER");
  int mode = 0;  // ablation: 1 = skip matvec dot, 2 = barriers only
  if (const char* e = getenv("KFAC_SYTRD_MODE")) mode = atoi(e);
  const int wgs_alloc = wgs;
  auto opts = As.options();
  auto W = at::empty({b, 64L, (long)n}, opts);
  auto E = at::zeros({b, (long)n}, opts);
  auto tau = at::zeros({b, (long)n}, opts);
  auto scratch = at::empty({b, (long)wgs_alloc, 130L}, opts);
  auto syncc = at::zeros({b}, opts.dtype(at::kInt));
  auto status = at::zeros({b}, opts.dtype(at::kInt));
  auto stream = c10::hip::getCurrentHIPStream();
  ROCBLASCHECK(rocblas_set_stream(g_main_handle, stream.stream()));
  const float one = 1.f, neg1 = -1.f;
  float* Ap = As.data_ptr<float>();
  float* Wp = W.data_ptr<float>();
  for (int j0 = 0; j0 < n - 1; j0 += 64) {
    const int ib = std::min(64, n - 1 - j0);
    const int wg = (n - j0 + R - 1) / R;
    sytrd_panels_device(As, W, E, tau, scratch, syncc, status, j0, ib,
                        R, wgs_alloc, wg, mode);
    const int t = j0 + ib;
    const int M = n - t;
    if (M > 0) {
      // row-major: A[t:, t:] -= V2^T W2 + W2^T V2 with V2 = A[j0:t, t:]
      // (reflector rows), W2 = W[0:ib, t:].  In rocBLAS column-major
      // terms (buffers viewed as M x ib with ld = n):
      // C -= P Q^T + Q P^T, C symmetric so the transposed view is fine.
      float* P = Ap + (size_t)j0 * n + t;      // V2, ld n, stride n*n
      float* Q = Wp + t;                       // W2, ld n, stride 64*n
      float* C = Ap + (size_t)t * n + t;       // trailing, ld n
      ROCBLASCHECK(rocblas_sgemm_strided_batched(
          g_main_handle, rocblas_operation_none,
          rocblas_operation_transpose, M, M, ib, &neg1, P, n,
          (rocblas_stride)n * n, Q, n, (rocblas_stride)64 * n, &one, C,
          n, (rocblas_stride)n * n, b));
      ROCBLASCHECK(rocblas_sgemm_strided_batched(
          g_main_handle, rocblas_operation_none,
          rocblas_operation_transpose, M, M, ib, &neg1, Q, n,
          (rocblas_stride)64 * n, P, n, (rocblas_stride)n * n, &one, C,
          n, (rocblas_stride)n * n, b));
    }
  }
  return {E, tau, status};
}

// Tridiagonal divide-and-conquer eigensolve on a pool slot (for
// overlapping the per-matrix stedc stage of the custom sytrd path).
// D (n,) is overwritten with ascending eigenvalues; returns (C, info)
// with C the eigenvector matrix, column-major in the buffer.
std::vector<torch::Tensor> stedc_slot_(torch::Tensor D, torch::Tensor E,
                                       int slot) {
  ensure_init();
  const long n = D.numel();
  TORCH_CHECK(D.is_cuda() && E.is_cuda() && E.numel() >= n - 1 &&
                  D.is_contiguous() && E.is_contiguous(),
              "stedc_slot_: contiguous device D/E required");
  auto C = at::eye(n, D.options());
  auto info = at::empty({1}, D.options().dtype(at::kInt));
  auto stream = c10::hip::getCurrentHIPStream();
  rocblas_handle h = g_main_handle;
  if (slot >= 0) {
    Slot& s = g_pool[slot % POOL];
    HIPCHECK(hipEventRecord(g_acq_event, stream.stream()));
    HIPCHECK(hipStreamWaitEvent(s.stream, g_acq_event, 0));
    record_on(D, s.stream);
    record_on(E, s.stream);
    record_on(C, s.stream);
    record_on(info, s.stream);
    h = s.handle;
  } else {
    ROCBLASCHECK(rocblas_set_stream(g_main_handle, stream.stream()));
  }
  ROCBLASCHECK(rocsolver_sstedc(h, rocblas_evect_original,
                                (rocblas_int)n, D.data_ptr<float>(),
                                E.data_ptr<float>(), C.data_ptr<float>(),
                                (rocblas_int)n, info.data_ptr<int>()));
  return {C, info};
}

std::vector<torch::Tensor> stedc_(torch::Tensor D, torch::Tensor E) {
  ensure_init();
  const long n = D.numel();
  TORCH_CHECK(D.is_cuda() && E.is_cuda() && E.numel() >= n - 1,
              "stedc_: device D/E required");
  auto C = at::eye(n, D.options());
  auto info = at::empty({1}, D.options().dtype(at::kInt));
  auto stream = c10::hip::getCurrentHIPStream();
  ROCBLASCHECK(rocblas_set_stream(g_main_handle, stream.stream()));
  ROCBLASCHECK(rocsolver_sstedc(g_main_handle, rocblas_evect_original,
                                (rocblas_int)n, D.data_ptr<float>(),
                                E.data_ptr<float>(), C.data_ptr<float>(),
                                (rocblas_int)n, info.data_ptr<int>()));
  return {C, info};
}

void ormtr_(torch::Tensor A, torch::Tensor tau, torch::Tensor C) {
  ensure_init();
  const long n = A.size(0);
  TORCH_CHECK(C.size(0) == n && C.size(1) == n && C.is_contiguous(),
              "ormtr_: C must be (n, n) contiguous");
  auto stream = c10::hip::getCurrentHIPStream();
  ROCBLASCHECK(rocblas_set_stream(g_main_handle, stream.stream()));
  ROCBLASCHECK(rocsolver_sormtr(
      g_main_handle, rocblas_side_left, rocblas_fill_lower,
      rocblas_operation_none, (rocblas_int)n, (rocblas_int)n,
      A.data_ptr<float>(), (rocblas_int)n, tau.data_ptr<float>(),
      C.data_ptr<float>(), (rocblas_int)n));
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "async/batched rocSOLVER eigensolves for K-FAC (MI355X)";
  m.def("syevdj_batched_", &syevdj_batched_,
        "in-place batched symmetric eigensolve of (b,n,n); returns "
        "(W, info); eigenvectors left row-major-transposed in input; "
        "slot >= 0 issues on that pool stream (join_pool_ after)",
        py::arg("As"), py::arg("slot") = -1);
  m.def("syevd_batched_", &syevd_batched_,
        "in-place strided-batched divide-and-conquer eigensolve; same "
        "contract as syevdj_batched_", py::arg("As"), py::arg("slot") = -1);
  m.def("join_pool_", &join_pool_,
        "torch current stream waits on all pool streams");
  m.def("sytrd_", &sytrd_, "in-place tridiagonalization; returns D,E,tau");
  m.def("sytrd_batched_custom_", &sytrd_batched_custom_,
        "hand-written batched blocked tridiagonalization (persistent "
        "panel kernel + strided-batched trailing GEMMs); in-place; "
        "returns (E, tau, status)");
  m.def("stedc_", &stedc_,
        "tridiagonal D&C eigensolve; returns (C eigvec col-major, info)");
  m.def("stedc_slot_", &stedc_slot_,
        "tridiagonal D&C eigensolve on a pool slot (join_pool_ after); "
        "returns (C, info)", py::arg("D"), py::arg("E"),
        py::arg("slot") = -1);
  m.def("ormtr_", &ormtr_,
        "C <- Q C with Q from sytrd reflectors (in-place)");
  m.def("syevd_pool_", &syevd_pool_,
        "in-place pool-stream-overlapped eigensolves of mixed sizes; "
        "returns [W..., info]");
  m.def("potri_pool_", &potri_pool_,
        "in-place pool-stream-overlapped Cholesky inverse (lower) of "
        "mixed sizes; returns [info]");
  m.def("potri_batched_", &potri_batched_,
        "in-place strided-batched Cholesky inverse of (b,n,n); "
        "returns [info]", py::arg("As"), py::arg("slot") = -1);
}
