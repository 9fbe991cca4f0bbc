// MI355X (gfx950 / CDNA4) kernels for the SAC training framework.
//
// Hand-written HIP — no CUDA shims, no hipify, gfx950 only.  The workload
// is latency-bound (models are sub-MB, batches 64..4096), so the design
// goals are: (1) MFMA matrix cores for every GEMM-shaped op with
// LDS-staged tiles, (2) maximal fusion — the tanh-Gaussian head, the
// Bellman backup + twin-min + MSE (+ gradient seeds), the whole-module
// polyak and Adam each run as ONE kernel, (3) every kernel is
// hipGraph-capture-safe: no host sync, RNG/step counters live in device
// memory and are bumped by tiny predecessor kernels on the same stream.
//
// Numerics contracts mirror the reference implementation:
//   * linear+relu trunk        — reference networks/linear.py:32-35
//   * tanh-Gaussian head       — reference networks/linear.py:37-53
//   * Bellman backup/twin MSE  — reference sac/algorithm.py:46-74
//   * policy loss              — reference sac/algorithm.py:30-43
//   * polyak / Adam            — reference sac/algorithm.py:77-81, main.py:94
//   * replay sample+gather     — reference buffer/replay_buffer.py:45-54
//     (with replacement — deviation documented in buffer/replay.py)

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <cmath>
#include <vector>

#define DEVINL __device__ __forceinline__

namespace {

constexpr int WAVE = 64;

// ---------------------------------------------------------------------------
// Philox4x32-10 counter-based RNG (device-side, graph-replay-safe)
// ---------------------------------------------------------------------------

struct Philox4 {
  uint32_t x, y, z, w;
};

DEVINL uint32_t mulhilo(uint32_t a, uint32_t b, uint32_t* hi) {
  uint64_t p = (uint64_t)a * (uint64_t)b;
  *hi = (uint32_t)(p >> 32);
  return (uint32_t)p;
}

DEVINL Philox4 philox4(uint64_t seed, uint64_t ctr_hi, uint64_t ctr_lo) {
  uint32_t c0 = (uint32_t)ctr_lo, c1 = (uint32_t)(ctr_lo >> 32);
  uint32_t c2 = (uint32_t)ctr_hi, c3 = (uint32_t)(ctr_hi >> 32);
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
  const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  const uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    uint32_t hi0, hi1;
    uint32_t lo0 = mulhilo(M0, c0, &hi0);
    uint32_t lo1 = mulhilo(M1, c2, &hi1);
    uint32_t n0 = hi1 ^ c1 ^ k0;
    uint32_t n1 = lo1;
    uint32_t n2 = hi0 ^ c3 ^ k1;
    uint32_t n3 = lo0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += W0; k1 += W1;
  }
  return {c0, c1, c2, c3};
}

// ---------------------------------------------------------------------------
// Counter bump (predecessor kernel for graph-safe RNG / Adam step)
// ---------------------------------------------------------------------------

__global__ void bump_counter_kernel(int64_t* ctr) { ++ctr[0]; }

// ---------------------------------------------------------------------------
// MFMA GEMM: Y[M,N] = X[M,K] @ W[N,K]^T (+bias) (+ReLU)
//
// fp32 mode: v_mfma_f32_16x16x4_f32 — exact f32 at the f32 vector rate.
// bf16 mode: v_mfma_f32_16x16x32_bf16 — inputs rounded to bf16 on LDS
//   stage (fp32 master weights stay in HBM), fp32 accumulate.
//
// Block = 256 threads = 4 waves (2x2), wave tile 32x32 (2x2 16x16 frags),
// block tile 64x64.  LDS rows padded to dodge bank conflicts (§2/G4 of
// the CDNA4 guide).  Shapes here are tiny (K,N <= 512), so tiles are
// sized for occupancy at small M/N, not for peak PF.
// ---------------------------------------------------------------------------

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

constexpr int BM = 64, BN = 64;
constexpr int BKF = 16;   // K-step fp32 (4 mfma of K=4)
constexpr int BKB = 32;   // K-step bf16 (1 mfma of K=32)
constexpr int LDF = BKF + 1;   // fp32 LDS row stride (17 dwords)
constexpr int LDB = BKB + 8;   // bf16 LDS row stride (40 halves = 20 dwords)

// OPA selects how the "A/B" operand is addressed from global memory:
//  N  : T[row, k]  (row-major over k, leading dim ldk)
//  T  : T[k, row]  (transposed — used for dy^T in wgrad)
enum class Op { N, T };

template <Op OP>
DEVINL float gload(const float* p, int row, int k, int nrows, int nk, int ld) {
  if constexpr (OP == Op::N) {
    return (row < nrows && k < nk) ? p[(int64_t)row * ld + k] : 0.f;
  } else {
    return (row < nrows && k < nk) ? p[(int64_t)k * ld + row] : 0.f;
  }
}

// Generic MFMA GEMM core, fp32 LDS.
// Computes acc = A_tile @ B_tile^T over K with A[M,K] via OPA, B[N,K] via
// OPB; epilogue left to caller via functor EPI(row, col, val).
template <Op OPA, Op OPB, bool BF16, typename Epi>
DEVINL void gemm_core(const float* __restrict__ A, const float* __restrict__ B,
                      int M, int N, int K, int lda, int ldb, Epi epi) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wrow = (wid >> 1) * 32;
  const int wcol = (wid & 1) * 32;
  const int bm0 = blockIdx.x * BM;
  const int bn0 = blockIdx.y * BN;

  constexpr int BK = BF16 ? BKB : BKF;
  __shared__ __attribute__((aligned(16))) char smem[
      BF16 ? (2 * 64 * LDB * 2) : (2 * 64 * LDF * 4)];

  f32x4 acc[2][2] = {};

  for (int k0 = 0; k0 < K; k0 += BK) {
    // ---- stage A[64][BK], B[64][BK] into LDS (predicated, zero-pad) ----
    if constexpr (!BF16) {
      float* xs = (float*)smem;                 // [64][LDF]
      float* ws = xs + 64 * LDF;                // [64][LDF]
      // 256 threads x 4 elems = 64x16
      const int row = tid & 63;
      const int c0 = (tid >> 6) * 4;
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        int c = c0 + e;
        xs[row * LDF + c] = gload<OPA>(A, bm0 + row, k0 + c, M, K, lda);
        ws[row * LDF + c] = gload<OPB>(B, bn0 + row, k0 + c, N, K, ldb);
      }
    } else {
      __bf16* xs = (__bf16*)smem;               // [64][LDB]
      __bf16* ws = xs + 64 * LDB;
      // 256 threads x 8 elems = 64x32
      const int row = tid & 63;
      const int c0 = (tid >> 6) * 8;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        int c = c0 + e;
        xs[row * LDB + c] = (__bf16)gload<OPA>(A, bm0 + row, k0 + c, M, K, lda);
        ws[row * LDB + c] = (__bf16)gload<OPB>(B, bn0 + row, k0 + c, N, K, ldb);
      }
    }
    __syncthreads();

    // ---- MFMA over the K-step ----
    if constexpr (!BF16) {
      const float* xs = (const float*)smem;
      const float* ws = xs + 64 * LDF;
      const int arow = lane & 15;      // fragment row
      const int akl = lane >> 4;       // fragment k sub-lane
#pragma unroll
      for (int kk = 0; kk < BKF; kk += 4) {
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
          float a = xs[(wrow + mi * 16 + arow) * LDF + kk + akl];
#pragma unroll
          for (int ni = 0; ni < 2; ++ni) {
            float b = ws[(wcol + ni * 16 + arow) * LDF + kk + akl];
            acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                a, b, acc[mi][ni], 0, 0, 0);
          }
        }
      }
    } else {
      const __bf16* xs = (const __bf16*)smem;
      const __bf16* ws = xs + 64 * LDB;
      const int arow = lane & 15;
      const int ak0 = (lane >> 4) * 8;
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        bf16x8 a = *(const bf16x8*)&xs[(wrow + mi * 16 + arow) * LDB + ak0];
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          bf16x8 b = *(const bf16x8*)&ws[(wcol + ni * 16 + arow) * LDB + ak0];
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, b, acc[mi][ni], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: C/D frag mapping col=lane&15, row=(lane>>4)*4+reg ----
  const int crow = (lane >> 4) * 4;
  const int ccol = lane & 15;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int grow = bm0 + wrow + mi * 16 + crow + r;
        int gcol = bn0 + wcol + ni * 16 + ccol;
        if (grow < M && gcol < N) epi(grow, gcol, acc[mi][ni][r]);
      }
}

template <bool RELU, bool BF16>
__global__ __launch_bounds__(256)
void linear_fwd_kernel(const float* __restrict__ X, const float* __restrict__ W,
                       const float* __restrict__ bias, float* __restrict__ Y,
                       int M, int N, int K) {
  gemm_core<Op::N, Op::N, BF16>(X, W, M, N, K, K, K,
      [&](int row, int col, float v) {
        v += bias ? bias[col] : 0.f;
        if constexpr (RELU) v = fmaxf(v, 0.f);
        Y[(int64_t)row * N + col] = v;
      });
}

// dX[M,K] = dY_eff[M,N] @ W[N,K];  dY_eff = dY * (Y > 0) when RELU.
// GEMM: A = dY_eff (M x N, op N over j), B = W^T view: B[k,j] = W[j,k]
// -> use OPB = T with leading dim K (W stored [N,K], element W[j,k] at
// j*K + k; we need B tile rows indexed by k: B[krow, j] = W[j, krow]).
template <bool RELU, bool BF16>
__global__ __launch_bounds__(256)
void linear_dgrad_kernel(const float* __restrict__ dY,
                         const float* __restrict__ Y,
                         const float* __restrict__ W,
                         float* __restrict__ dX, int M, int N, int K) {
  // Treat as GEMM (M x K) reducing over N: A[i,j] = dY_eff[i,j] (ld N),
  // B[k,j] = W[j,k] -> OPB=T with ld K.
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wrow = (wid >> 1) * 32;
  const int wcol = (wid & 1) * 32;
  const int bm0 = blockIdx.x * BM;
  const int bn0 = blockIdx.y * BN;   // over K dim of dX

  constexpr int BK = BF16 ? BKB : BKF;
  __shared__ __attribute__((aligned(16))) char smem[
      BF16 ? (2 * 64 * LDB * 2) : (2 * 64 * LDF * 4)];
  f32x4 acc[2][2] = {};

  for (int j0 = 0; j0 < N; j0 += BK) {
    if constexpr (!BF16) {
      float* as = (float*)smem;
      float* bs = as + 64 * LDF;
      const int row = tid & 63;
      const int c0 = (tid >> 6) * 4;
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        int j = j0 + c0 + e;
        float dy = (bm0 + row < M && j < N)
                       ? dY[(int64_t)(bm0 + row) * N + j] : 0.f;
        if constexpr (RELU) {
          float y = (bm0 + row < M && j < N)
                        ? Y[(int64_t)(bm0 + row) * N + j] : 0.f;
          dy = y > 0.f ? dy : 0.f;
        }
        as[row * LDF + c0 + e] = dy;
        bs[row * LDF + c0 + e] = (bn0 + row < K && j < N)
                                     ? W[(int64_t)j * K + bn0 + row] : 0.f;
      }
    } else {
      __bf16* as = (__bf16*)smem;
      __bf16* bs = as + 64 * LDB;
      const int row = tid & 63;
      const int c0 = (tid >> 6) * 8;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        int j = j0 + c0 + e;
        float dy = (bm0 + row < M && j < N)
                       ? dY[(int64_t)(bm0 + row) * N + j] : 0.f;
        if constexpr (RELU) {
          float y = (bm0 + row < M && j < N)
                        ? Y[(int64_t)(bm0 + row) * N + j] : 0.f;
          dy = y > 0.f ? dy : 0.f;
        }
        as[row * LDB + c0 + e] = (__bf16)dy;
        bs[row * LDB + c0 + e] = (__bf16)((bn0 + row < K && j < N)
                                     ? W[(int64_t)j * K + bn0 + row] : 0.f);
      }
    }
    __syncthreads();

    if constexpr (!BF16) {
      const float* as = (const float*)smem;
      const float* bs = as + 64 * LDF;
      const int arow = lane & 15, akl = lane >> 4;
#pragma unroll
      for (int kk = 0; kk < BKF; kk += 4)
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
          float a = as[(wrow + mi * 16 + arow) * LDF + kk + akl];
#pragma unroll
          for (int ni = 0; ni < 2; ++ni) {
            float b = bs[(wcol + ni * 16 + arow) * LDF + kk + akl];
            acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                a, b, acc[mi][ni], 0, 0, 0);
          }
        }
    } else {
      const __bf16* as = (const __bf16*)smem;
      const __bf16* bs = as + 64 * LDB;
      const int arow = lane & 15, ak0 = (lane >> 4) * 8;
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        bf16x8 a = *(const bf16x8*)&as[(wrow + mi * 16 + arow) * LDB + ak0];
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          bf16x8 b = *(const bf16x8*)&bs[(wcol + ni * 16 + arow) * LDB + ak0];
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, b, acc[mi][ni], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  const int crow = (lane >> 4) * 4, ccol = lane & 15;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int grow = bm0 + wrow + mi * 16 + crow + r;
        int gcol = bn0 + wcol + ni * 16 + ccol;
        if (grow < M && gcol < K)
          dX[(int64_t)grow * K + gcol] = acc[mi][ni][r];
      }
}

// dW[N,K] = dY_eff^T[N,M] @ X[M,K]; db[N] = sum_i dY_eff[i,N]
// A[n,i] = dY_eff[i,n] -> OPA=T (ld N); B tile rows are k: B[k... wait —
// GEMM form: rows of output = n, cols = k, reduce over i (batch M).
template <bool RELU, bool BF16>
__global__ __launch_bounds__(256)
void linear_wgrad_kernel(const float* __restrict__ dY,
                         const float* __restrict__ Y,
                         const float* __restrict__ X,
                         float* __restrict__ dW, float* __restrict__ db,
                         int M, int N, int K) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wrow = (wid >> 1) * 32;
  const int wcol = (wid & 1) * 32;
  const int bn0 = blockIdx.x * BM;   // over N (output rows)
  const int bk0 = blockIdx.y * BN;   // over K (output cols)

  constexpr int BK = BF16 ? BKB : BKF;
  __shared__ __attribute__((aligned(16))) char smem[
      BF16 ? (2 * 64 * LDB * 2) : (2 * 64 * LDF * 4)];
  __shared__ float dbs[64];
  f32x4 acc[2][2] = {};
  float db_acc = 0.f;   // thread-local; reduced at the end (k-tile 0 only)

  for (int i0 = 0; i0 < M; i0 += BK) {
    if constexpr (!BF16) {
      float* as = (float*)smem;          // as[n][i] tile
      float* bs = as + 64 * LDF;         // bs[k][i] tile
      const int row = tid & 63;
      const int c0 = (tid >> 6) * 4;
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        int i = i0 + c0 + e;
        float dy = (bn0 + row < N && i < M)
                       ? dY[(int64_t)i * N + bn0 + row] : 0.f;
        if constexpr (RELU) {
          float y = (bn0 + row < N && i < M)
                        ? Y[(int64_t)i * N + bn0 + row] : 0.f;
          dy = y > 0.f ? dy : 0.f;
        }
        as[row * LDF + c0 + e] = dy;
        db_acc += dy;
        bs[row * LDF + c0 + e] = (bk0 + row < K && i < M)
                                     ? X[(int64_t)i * K + bk0 + row] : 0.f;
      }
    } else {
      __bf16* as = (__bf16*)smem;
      __bf16* bs = as + 64 * LDB;
      const int row = tid & 63;
      const int c0 = (tid >> 6) * 8;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        int i = i0 + c0 + e;
        float dy = (bn0 + row < N && i < M)
                       ? dY[(int64_t)i * N + bn0 + row] : 0.f;
        if constexpr (RELU) {
          float y = (bn0 + row < N && i < M)
                        ? Y[(int64_t)i * N + bn0 + row] : 0.f;
          dy = y > 0.f ? dy : 0.f;
        }
        as[row * LDB + c0 + e] = (__bf16)dy;
        db_acc += dy;
        bs[row * LDB + c0 + e] = (__bf16)((bk0 + row < K && i < M)
                                     ? X[(int64_t)i * K + bk0 + row] : 0.f);
      }
    }
    __syncthreads();

    if constexpr (!BF16) {
      const float* as = (const float*)smem;
      const float* bs = as + 64 * LDF;
      const int arow = lane & 15, akl = lane >> 4;
#pragma unroll
      for (int kk = 0; kk < BKF; kk += 4)
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
          float a = as[(wrow + mi * 16 + arow) * LDF + kk + akl];
#pragma unroll
          for (int ni = 0; ni < 2; ++ni) {
            float b = bs[(wcol + ni * 16 + arow) * LDF + kk + akl];
            acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                a, b, acc[mi][ni], 0, 0, 0);
          }
        }
    } else {
      const __bf16* as = (const __bf16*)smem;
      const __bf16* bs = as + 64 * LDB;
      const int arow = lane & 15, ak0 = (lane >> 4) * 8;
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        bf16x8 a = *(const bf16x8*)&as[(wrow + mi * 16 + arow) * LDB + ak0];
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          bf16x8 b = *(const bf16x8*)&bs[(wcol + ni * 16 + arow) * LDB + ak0];
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, b, acc[mi][ni], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  const int crow = (lane >> 4) * 4, ccol = lane & 15;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int gn = bn0 + wrow + mi * 16 + crow + r;
        int gk = bk0 + wcol + ni * 16 + ccol;
        if (gn < N && gk < K) dW[(int64_t)gn * K + gk] = acc[mi][ni][r];
      }

  // db: each LDS-stage row of `as` is one n; thread (row=tid&63) summed
  // dy over the i-columns it loaded -> reduce the 4 thread-copies per row.
  if (db != nullptr && blockIdx.y == 0) {
    const int row = tid & 63;
    // 4 threads share each row (c0 = 0,4,8,12 / 0,8,... in bf16): reduce
    // via LDS.
    if (tid < 64) dbs[tid] = 0.f;
    __syncthreads();
    atomicAdd(&dbs[row], db_acc);
    __syncthreads();
    if (tid < 64 && bn0 + tid < N) db[bn0 + tid] = dbs[tid];
  }
}

// ---------------------------------------------------------------------------
// Fused tanh-Gaussian head (reference networks/linear.py:37-53)
// ---------------------------------------------------------------------------

constexpr float kLog2Pi = 1.8378770664093453f;  // log(2*pi)
constexpr float kTwoLog2 = 1.3862943611198906f; // 2*log(2)

DEVINL float softplus_neg2(float prob) {
  // softplus(-2*prob) computed stably
  float x = -2.f * prob;
  return x > 20.f ? x : log1pf(expf(fminf(x, 20.f)));
}

// One wave per row (act_dim <= 64 covers the whole suite: 1..56).
// grid.x = B, block = 64.
template <bool DET, bool WITH_LOGP>
__global__ __launch_bounds__(64)
void tanh_gauss_fwd_kernel(const float* __restrict__ mu,
                           const float* __restrict__ log_std,
                           const float* __restrict__ eps,
                           float* __restrict__ pi, float* __restrict__ logp,
                           float* __restrict__ prob_out,
                           float* __restrict__ ls_c_out,
                           int B, int A, float act_limit, float lo, float hi) {
  const int b = blockIdx.x;
  const int a = threadIdx.x;
  float acc = 0.f;
  if (a < A) {
    const int64_t i = (int64_t)b * A + a;
    float m = mu[i];
    float ls = fminf(fmaxf(log_std[i], lo), hi);
    float std = expf(ls);
    float prob = DET ? m : m + std * eps[i];
    pi[i] = tanhf(prob) * act_limit;
    prob_out[i] = prob;
    ls_c_out[i] = ls;
    if constexpr (WITH_LOGP) {
      float e = (prob - m) / std;
      float gauss = -0.5f * e * e - ls - 0.5f * kLog2Pi;
      float corr = kTwoLog2 - prob - softplus_neg2(prob);
      acc = gauss - corr;
    }
  }
  if constexpr (WITH_LOGP) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      acc += __shfl_down(acc, off);
    if (a == 0) logp[b] = acc;
  }
}

// backward: dmu, dlog_std from dpi [B,A] and dlogp [B]
template <bool DET>
__global__ __launch_bounds__(64)
void tanh_gauss_bwd_kernel(const float* __restrict__ dpi,
                           const float* __restrict__ dlogp,
                           const float* __restrict__ mu,
                           const float* __restrict__ log_std,
                           const float* __restrict__ eps,
                           const float* __restrict__ prob,
                           const float* __restrict__ ls_c,
                           float* __restrict__ dmu,
                           float* __restrict__ dls,
                           int B, int A, float act_limit, float lo, float hi,
                           bool with_logp) {
  const int b = blockIdx.x;
  const int a = threadIdx.x;
  if (a >= A) return;
  const int64_t i = (int64_t)b * A + a;
  float ls = ls_c[i];
  float std = expf(ls);
  float p = prob[i];
  float t = tanhf(p);
  float se = DET ? 0.f : std * eps[i];    // std * eps = prob - mu
  float dp = dpi[i] * act_limit * (1.f - t * t);
  float dl = with_logp ? dlogp[b] : 0.f;

  // d logp / dmu = tanh(prob); d logp / dls = se*tanh(prob) - 1
  float g_mu = dp + dl * t;
  float g_ls = dp * se + dl * (se * t - 1.f);
  // clip mask (torch.clamp backward: pass-through inside [lo, hi])
  float raw = log_std[i];
  float mask = (raw >= lo && raw <= hi) ? 1.f : 0.f;
  dmu[i] = g_mu;
  dls[i] = g_ls * mask;
}

// ---------------------------------------------------------------------------
// Fused SAC losses (+ gradient seeds)
// ---------------------------------------------------------------------------

// loss_q = mse(q1, backup) + mse(q2, backup);
// backup = scale*r + gamma*(1-d)*(min(q1t,q2t) - alpha*logp)
// dq1 = 2*(q1-backup)/B, dq2 likewise.  grid-stride, block 256.
__global__ __launch_bounds__(256)
void sac_q_loss_kernel(const float* __restrict__ q1,
                       const float* __restrict__ q2,
                       const float* __restrict__ q1t,
                       const float* __restrict__ q2t,
                       const float* __restrict__ logp,
                       const float* __restrict__ rew,
                       const float* __restrict__ done,
                       float* __restrict__ loss,
                       float* __restrict__ dq1, float* __restrict__ dq2,
                       int B, float alpha, float gamma, float scale) {
  __shared__ float red[4];
  float acc = 0.f;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < B;
       i += gridDim.x * blockDim.x) {
    float backup = scale * rew[i] + gamma * (1.f - done[i]) *
                       (fminf(q1t[i], q2t[i]) - alpha * logp[i]);
    float e1 = q1[i] - backup;
    float e2 = q2[i] - backup;
    acc += e1 * e1 + e2 * e2;
    dq1[i] = 2.f * e1 / B;
    dq2[i] = 2.f * e2 / B;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = red[0] + red[1] + red[2] + red[3];
    atomicAdd(loss, s / B);
  }
}

// loss_pi = mean(alpha*logp - min(q1,q2));
// dq on the min branch (0.5/0.5 at exact ties to match torch.minimum).
__global__ __launch_bounds__(256)
void sac_pi_loss_kernel(const float* __restrict__ q1,
                        const float* __restrict__ q2,
                        const float* __restrict__ logp,
                        float* __restrict__ loss,
                        float* __restrict__ dq1, float* __restrict__ dq2,
                        float* __restrict__ dlogp,
                        int B, float alpha) {
  __shared__ float red[4];
  float acc = 0.f;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < B;
       i += gridDim.x * blockDim.x) {
    float a = q1[i], b = q2[i];
    acc += alpha * logp[i] - fminf(a, b);
    float g1, g2;
    if (a < b)      { g1 = -1.f; g2 = 0.f; }
    else if (b < a) { g1 = 0.f;  g2 = -1.f; }
    else            { g1 = -0.5f; g2 = -0.5f; }
    dq1[i] = g1 / B;
    dq2[i] = g2 / B;
    dlogp[i] = alpha / B;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = red[0] + red[1] + red[2] + red[3];
    atomicAdd(loss, s / B);
  }
}

// ---------------------------------------------------------------------------
// Flat-buffer maintenance: polyak / Adam (one kernel per module)
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256)
void polyak_kernel(float* __restrict__ targ, const float* __restrict__ src,
                   int64_t n, float rho) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
       i0 < n; i0 += stride) {
    if (i0 + 3 < n) {
      float4 t = *(float4*)&targ[i0];
      float4 s = *(const float4*)&src[i0];
      t.x = rho * t.x + (1.f - rho) * s.x;
      t.y = rho * t.y + (1.f - rho) * s.y;
      t.z = rho * t.z + (1.f - rho) * s.z;
      t.w = rho * t.w + (1.f - rho) * s.w;
      *(float4*)&targ[i0] = t;
    } else {
      for (int64_t i = i0; i < n; ++i)
        targ[i] = rho * targ[i] + (1.f - rho) * src[i];
    }
  }
}

// Reads step AFTER the bump kernel incremented it (same-stream order).
__global__ __launch_bounds__(256)
void adam_kernel(float* __restrict__ p, const float* __restrict__ g,
                 float* __restrict__ m, float* __restrict__ v,
                 const int64_t* __restrict__ step, int64_t n,
                 float lr, float b1, float b2, float eps, float wd) {
  const float t = (float)step[0];
  const float bc1 = 1.f - powf(b1, t);
  const float bc2 = 1.f - powf(b2, t);
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float gi = g[i] + wd * p[i];
    float mi = b1 * m[i] + (1.f - b1) * gi;
    float vi = b2 * v[i] + (1.f - b2) * gi * gi;
    m[i] = mi;
    v[i] = vi;
    p[i] -= lr / bc1 * mi / (sqrtf(vi / bc2) + eps);
  }
}

// ---------------------------------------------------------------------------
// Replay sample + gather (Philox, with replacement)
// ---------------------------------------------------------------------------

// grid.x = B (one block per sampled row); all threads of a block copy the
// row's fields.  idx_j = philox(seed, ctr, j) % size.
__global__ __launch_bounds__(256)
void replay_gather_kernel(const float* __restrict__ state,
                          const float* __restrict__ act,
                          const float* __restrict__ rew,
                          const float* __restrict__ nstate,
                          const float* __restrict__ done,
                          const int64_t* __restrict__ size_dev,
                          const int64_t* __restrict__ ctr,
                          uint64_t seed,
                          float* __restrict__ os, float* __restrict__ oa,
                          float* __restrict__ orew, float* __restrict__ ons,
                          float* __restrict__ od,
                          int obs_dim, int act_dim) {
  const int j = blockIdx.x;
  const uint64_t size = (uint64_t)size_dev[0];
  Philox4 r = philox4(seed, (uint64_t)ctr[0], (uint64_t)j);
  // 64-bit uniform to keep modulo bias negligible at 1e6 sizes
  uint64_t u = ((uint64_t)r.x << 32) | r.y;
  int64_t idx = (int64_t)(u % (size ? size : 1));

  const float* srow = state + (int64_t)idx * obs_dim;
  const float* nrow = nstate + (int64_t)idx * obs_dim;
  float* osr = os + (int64_t)j * obs_dim;
  float* onr = ons + (int64_t)j * obs_dim;
  for (int c = threadIdx.x; c < obs_dim; c += blockDim.x) {
    osr[c] = srow[c];
    onr[c] = nrow[c];
  }
  const float* arow = act + (int64_t)idx * act_dim;
  float* oar = oa + (int64_t)j * act_dim;
  for (int c = threadIdx.x; c < act_dim; c += blockDim.x) oar[c] = arow[c];
  if (threadIdx.x == 0) {
    orew[j] = rew[idx];
    od[j] = done[idx];
  }
}

// ---------------------------------------------------------------------------
// Visual replay gather: ONE kernel draws the Philox index and copies
// features / frames (with u8 -> f32 dequantization) / actions / rewards
// / done into the static batch — replaces ~18 aten index/copy/decode
// launches per update in the captured visual SAC graph.
// ---------------------------------------------------------------------------

template <bool QUANT>
__global__ __launch_bounds__(256)
void visual_gather_kernel(const float* __restrict__ feat,
                          const void* __restrict__ frames,
                          const float* __restrict__ nfeat,
                          const void* __restrict__ nframes,
                          const float* __restrict__ act,
                          const float* __restrict__ rew,
                          const float* __restrict__ done,
                          const int64_t* __restrict__ size_dev,
                          const int64_t* __restrict__ ctr, uint64_t seed,
                          float* __restrict__ of, float* __restrict__ ofr,
                          float* __restrict__ onf,
                          float* __restrict__ onfr,
                          float* __restrict__ oa,
                          float* __restrict__ orew,
                          float* __restrict__ od,
                          int feat_dim, int64_t frame_n, int act_dim) {
  // grid (B, SLICES): every block re-derives the Philox index for its
  // sample j; blockIdx.y partitions the frame copy so the whole gather
  // fills the chip instead of B workgroups
  const int j = blockIdx.x;
  const int slice = blockIdx.y;
  const int nslice = gridDim.y;
  const uint64_t size = (uint64_t)size_dev[0];
  Philox4 r = philox4(seed, (uint64_t)ctr[0], (uint64_t)j);
  uint64_t u = ((uint64_t)r.x << 32) | r.y;
  const int64_t idx = (int64_t)(u % (size ? size : 1));

  if (slice == 0) {
    for (int c = threadIdx.x; c < feat_dim; c += blockDim.x) {
      of[(int64_t)j * feat_dim + c] = feat[idx * feat_dim + c];
      onf[(int64_t)j * feat_dim + c] = nfeat[idx * feat_dim + c];
    }
    for (int c = threadIdx.x; c < act_dim; c += blockDim.x)
      oa[(int64_t)j * act_dim + c] = act[idx * act_dim + c];
    if (threadIdx.x == 0) {
      orew[j] = rew[idx];
      od[j] = done[idx];
    }
  }
  const int64_t chunk = (frame_n + nslice - 1) / nslice;
  const int64_t c_lo = (int64_t)slice * chunk;
  const int64_t c_hi = min(frame_n, c_lo + chunk);
  if constexpr (QUANT) {
    const uint8_t* f0 = (const uint8_t*)frames + idx * frame_n;
    const uint8_t* f1 = (const uint8_t*)nframes + idx * frame_n;
    for (int64_t c = c_lo + threadIdx.x; c < c_hi; c += blockDim.x) {
      ofr[(int64_t)j * frame_n + c] = (float)f0[c] / 127.5f - 1.0f;
      onfr[(int64_t)j * frame_n + c] = (float)f1[c] / 127.5f - 1.0f;
    }
  } else {
    const float* f0 = (const float*)frames + idx * frame_n;
    const float* f1 = (const float*)nframes + idx * frame_n;
    for (int64_t c = c_lo + threadIdx.x; c < c_hi; c += blockDim.x) {
      ofr[(int64_t)j * frame_n + c] = f0[c];
      onfr[(int64_t)j * frame_n + c] = f1[c];
    }
  }
}

// ---------------------------------------------------------------------------
// Philox standard-normal sampler (Box-Muller) — graph-replay-safe noise
// for the reparameterized policy sample (torch's RNG offset is host-side;
// this keeps the whole SAC update replayable with fresh noise).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256)
void philox_randn_kernel(float* __restrict__ out, int64_t n,
                         const int64_t* __restrict__ ctr, uint64_t seed) {
  const uint64_t c = (uint64_t)ctr[0];
  const int64_t i4 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i4 * 4 >= n) return;
  Philox4 r = philox4(seed, c, (uint64_t)i4);
  // Box-Muller on two uniform pairs
  const float TWO_PI = 6.283185307179586f;
  float u0 = (r.x + 1.f) * 2.3283064365386963e-10f;  // (0,1]
  float u1 = (r.y + 1.f) * 2.3283064365386963e-10f;
  float u2 = (r.z + 1.f) * 2.3283064365386963e-10f;
  float u3 = (r.w + 1.f) * 2.3283064365386963e-10f;
  float r0 = sqrtf(-2.f * logf(u0));
  float r1 = sqrtf(-2.f * logf(u2));
  float s0, c0, s1, c1;
  __sincosf(TWO_PI * u1, &s0, &c0);
  __sincosf(TWO_PI * u3, &s1, &c1);
  float vals[4] = {r0 * c0, r0 * s0, r1 * c1, r1 * s1};
  int64_t base = i4 * 4;
#pragma unroll
  for (int e = 0; e < 4; ++e)
    if (base + e < n) out[base + e] = vals[e];
}

// ---------------------------------------------------------------------------
// Host-side launchers / bindings
// ---------------------------------------------------------------------------

inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

#define CHECK_IN(x) \
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dtype() == torch::kFloat32, \
              #x " must be contiguous fp32 CUDA tensor")

bool g_bf16 = false;  // compute mode: bf16 MFMA inputs (fp32 master data)

void set_compute_bf16(bool on) { g_bf16 = on; }
bool get_compute_bf16() { return g_bf16; }

torch::Tensor linear_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                         bool relu) {
  CHECK_IN(x); CHECK_IN(w);
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "shape mismatch");
  auto y = torch::empty({M, N}, x.options());
  dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN);
  const float* bias = nullptr;
  if (b.defined() && b.numel() > 0) { CHECK_IN(b); bias = b.data_ptr<float>(); }
  auto launch = [&](auto relu_c, auto bf16_c) {
    hipLaunchKernelGGL(
        (linear_fwd_kernel<decltype(relu_c)::value, decltype(bf16_c)::value>),
        grid, dim3(256), 0, cur_stream(),
        x.data_ptr<float>(), w.data_ptr<float>(), bias, y.data_ptr<float>(),
        M, N, K);
  };
  if (relu) {
    if (g_bf16) launch(std::true_type{}, std::true_type{});
    else        launch(std::true_type{}, std::false_type{});
  } else {
    if (g_bf16) launch(std::false_type{}, std::true_type{});
    else        launch(std::false_type{}, std::false_type{});
  }
  return y;
}

std::vector<torch::Tensor> linear_bwd(torch::Tensor dy, torch::Tensor x,
                                      torch::Tensor w, torch::Tensor y,
                                      bool relu, bool need_dx) {
  CHECK_IN(dy); CHECK_IN(x); CHECK_IN(w); CHECK_IN(y);
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  auto dw = torch::empty_like(w);
  auto db = torch::empty({N}, w.options());
  torch::Tensor dx;
  auto s = cur_stream();

  auto launch_wgrad = [&](auto relu_c, auto bf16_c) {
    dim3 grid((N + BM - 1) / BM, (K + BN - 1) / BN);
    hipLaunchKernelGGL(
        (linear_wgrad_kernel<decltype(relu_c)::value, decltype(bf16_c)::value>),
        grid, dim3(256), 0, s,
        dy.data_ptr<float>(), y.data_ptr<float>(), x.data_ptr<float>(),
        dw.data_ptr<float>(), db.data_ptr<float>(), M, N, K);
  };
  auto launch_dgrad = [&](auto relu_c, auto bf16_c) {
    dim3 grid((M + BM - 1) / BM, (K + BN - 1) / BN);
    hipLaunchKernelGGL(
        (linear_dgrad_kernel<decltype(relu_c)::value, decltype(bf16_c)::value>),
        grid, dim3(256), 0, s,
        dy.data_ptr<float>(), y.data_ptr<float>(), w.data_ptr<float>(),
        dx.data_ptr<float>(), M, N, K);
  };

#define DISPATCH(fn)                                        \
  if (relu) {                                               \
    if (g_bf16) fn(std::true_type{}, std::true_type{});     \
    else        fn(std::true_type{}, std::false_type{});    \
  } else {                                                  \
    if (g_bf16) fn(std::false_type{}, std::true_type{});    \
    else        fn(std::false_type{}, std::false_type{});   \
  }

  DISPATCH(launch_wgrad);
  if (need_dx) {
    dx = torch::empty_like(x);
    DISPATCH(launch_dgrad);
  }
#undef DISPATCH
  return {dx, dw, db};
}

std::vector<torch::Tensor> tanh_gauss_fwd(torch::Tensor mu,
                                          torch::Tensor log_std,
                                          torch::Tensor eps, double act_limit,
                                          double lo, double hi,
                                          bool deterministic,
                                          bool with_logprob) {
  CHECK_IN(mu); CHECK_IN(log_std); CHECK_IN(eps);
  const int B = mu.size(0), A = mu.size(1);
  TORCH_CHECK(A <= 64, "act_dim > 64 not supported by fused head");
  auto pi = torch::empty_like(mu);
  auto prob = torch::empty_like(mu);
  auto ls_c = torch::empty_like(mu);
  auto logp = torch::empty({B}, mu.options());
  auto s = cur_stream();
  auto launch = [&](auto det_c, auto wl_c) {
    hipLaunchKernelGGL(
        (tanh_gauss_fwd_kernel<decltype(det_c)::value, decltype(wl_c)::value>),
        dim3(B), dim3(64), 0, s,
        mu.data_ptr<float>(), log_std.data_ptr<float>(), eps.data_ptr<float>(),
        pi.data_ptr<float>(), logp.data_ptr<float>(), prob.data_ptr<float>(),
        ls_c.data_ptr<float>(), B, A, (float)act_limit, (float)lo, (float)hi);
  };
  if (deterministic) {
    if (with_logprob) launch(std::true_type{}, std::true_type{});
    else              launch(std::true_type{}, std::false_type{});
  } else {
    if (with_logprob) launch(std::false_type{}, std::true_type{});
    else              launch(std::false_type{}, std::false_type{});
  }
  return {pi, logp, prob, ls_c};
}

std::vector<torch::Tensor> tanh_gauss_bwd(torch::Tensor dpi,
                                          torch::Tensor dlogp,
                                          torch::Tensor mu,
                                          torch::Tensor log_std,
                                          torch::Tensor eps,
                                          torch::Tensor prob,
                                          torch::Tensor ls_c,
                                          double act_limit, double lo,
                                          double hi, bool deterministic,
                                          bool with_logprob) {
  CHECK_IN(dpi); CHECK_IN(mu);
  const int B = mu.size(0), A = mu.size(1);
  auto dmu = torch::empty_like(mu);
  auto dls = torch::empty_like(mu);
  auto s = cur_stream();
  auto launch = [&](auto det_c) {
    hipLaunchKernelGGL(
        (tanh_gauss_bwd_kernel<decltype(det_c)::value>),
        dim3(B), dim3(64), 0, s,
        dpi.data_ptr<float>(), dlogp.data_ptr<float>(), mu.data_ptr<float>(),
        log_std.data_ptr<float>(), eps.data_ptr<float>(),
        prob.data_ptr<float>(), ls_c.data_ptr<float>(),
        dmu.data_ptr<float>(), dls.data_ptr<float>(), B, A,
        (float)act_limit, (float)lo, (float)hi, with_logprob);
  };
  if (deterministic) launch(std::true_type{});
  else               launch(std::false_type{});
  return {dmu, dls};
}

std::vector<torch::Tensor> sac_q_loss_fwd(torch::Tensor q1, torch::Tensor q2,
                                          torch::Tensor q1t, torch::Tensor q2t,
                                          torch::Tensor logp, torch::Tensor rew,
                                          torch::Tensor done, double alpha,
                                          double gamma, double scale) {
  CHECK_IN(q1); CHECK_IN(q2);
  const int B = q1.size(0);
  auto loss = torch::zeros({}, q1.options());
  auto dq1 = torch::empty_like(q1);
  auto dq2 = torch::empty_like(q2);
  int blocks = std::min((B + 255) / 256, 64);
  hipLaunchKernelGGL(sac_q_loss_kernel, dim3(blocks), dim3(256), 0,
                     cur_stream(),
                     q1.data_ptr<float>(), q2.data_ptr<float>(),
                     q1t.data_ptr<float>(), q2t.data_ptr<float>(),
                     logp.data_ptr<float>(), rew.data_ptr<float>(),
                     done.data_ptr<float>(), loss.data_ptr<float>(),
                     dq1.data_ptr<float>(), dq2.data_ptr<float>(), B,
                     (float)alpha, (float)gamma, (float)scale);
  return {loss, dq1, dq2};
}

std::vector<torch::Tensor> sac_pi_loss_fwd(torch::Tensor q1, torch::Tensor q2,
                                           torch::Tensor logp, double alpha) {
  CHECK_IN(q1); CHECK_IN(q2); CHECK_IN(logp);
  const int B = q1.size(0);
  auto loss = torch::zeros({}, q1.options());
  auto dq1 = torch::empty_like(q1);
  auto dq2 = torch::empty_like(q2);
  auto dlogp = torch::empty_like(logp);
  int blocks = std::min((B + 255) / 256, 64);
  hipLaunchKernelGGL(sac_pi_loss_kernel, dim3(blocks), dim3(256), 0,
                     cur_stream(),
                     q1.data_ptr<float>(), q2.data_ptr<float>(),
                     logp.data_ptr<float>(), loss.data_ptr<float>(),
                     dq1.data_ptr<float>(), dq2.data_ptr<float>(),
                     dlogp.data_ptr<float>(), B, (float)alpha);
  return {loss, dq1, dq2, dlogp};
}

void polyak_(torch::Tensor targ, torch::Tensor src, double rho) {
  CHECK_IN(targ); CHECK_IN(src);
  int64_t n = targ.numel();
  int blocks = (int)std::min<int64_t>((n / 4 + 255) / 256 + 1, 1024);
  hipLaunchKernelGGL(polyak_kernel, dim3(blocks), dim3(256), 0, cur_stream(),
                     targ.data_ptr<float>(), src.data_ptr<float>(), n,
                     (float)rho);
}

void adam_step_(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, torch::Tensor step, double lr, double b1,
                double b2, double eps, double wd) {
  CHECK_IN(p); CHECK_IN(g); CHECK_IN(m); CHECK_IN(v);
  int64_t n = p.numel();
  auto s = cur_stream();
  hipLaunchKernelGGL(bump_counter_kernel, dim3(1), dim3(1), 0, s,
                     step.data_ptr<int64_t>());
  int blocks = (int)std::min<int64_t>((n + 255) / 256, 1024);
  hipLaunchKernelGGL(adam_kernel, dim3(blocks), dim3(256), 0, s,
                     p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(),
                     step.data_ptr<int64_t>(), n, (float)lr, (float)b1,
                     (float)b2, (float)eps, (float)wd);
}

void replay_sample_into(torch::Tensor state, torch::Tensor act,
                        torch::Tensor rew, torch::Tensor nstate,
                        torch::Tensor done, torch::Tensor size_dev,
                        torch::Tensor ctr, int64_t seed,
                        torch::Tensor os, torch::Tensor oa, torch::Tensor orew,
                        torch::Tensor ons, torch::Tensor od) {
  CHECK_IN(state); CHECK_IN(os);
  const int B = os.size(0);
  const int obs_dim = state.size(1);
  const int act_dim = act.size(1);
  auto s = cur_stream();
  hipLaunchKernelGGL(bump_counter_kernel, dim3(1), dim3(1), 0, s,
                     ctr.data_ptr<int64_t>());
  int threads = std::min(256, std::max(64, ((obs_dim + 63) / 64) * 64));
  hipLaunchKernelGGL(replay_gather_kernel, dim3(B), dim3(threads), 0, s,
                     state.data_ptr<float>(), act.data_ptr<float>(),
                     rew.data_ptr<float>(), nstate.data_ptr<float>(),
                     done.data_ptr<float>(), size_dev.data_ptr<int64_t>(),
                     ctr.data_ptr<int64_t>(), (uint64_t)seed,
                     os.data_ptr<float>(), oa.data_ptr<float>(),
                     orew.data_ptr<float>(), ons.data_ptr<float>(),
                     od.data_ptr<float>(), obs_dim, act_dim);
}

// One-kernel visual ring store: quantize both fp32 frames to u8 and
// write every field of the transition at ring slot `idx` — replaces the
// CPU-side encode chain + ~10 aten copies per env step.
template <bool QUANT>
__global__ __launch_bounds__(256)
void visual_store_kernel(const float* __restrict__ feat,
                         const float* __restrict__ frame,
                         const float* __restrict__ nfeat,
                         const float* __restrict__ nframe,
                         const float* __restrict__ act,
                         float rew, float done,
                         float* __restrict__ rfeat,
                         void* __restrict__ rframe,
                         float* __restrict__ rnfeat,
                         void* __restrict__ rnframe,
                         float* __restrict__ ract,
                         float* __restrict__ rrew,
                         float* __restrict__ rdone,
                         int64_t idx, int feat_dim, int64_t frame_n,
                         int act_dim) {
  const int slice = blockIdx.x;
  const int nslice = gridDim.x;
  if (slice == 0) {
    for (int c = threadIdx.x; c < feat_dim; c += blockDim.x) {
      rfeat[idx * feat_dim + c] = feat[c];
      rnfeat[idx * feat_dim + c] = nfeat[c];
    }
    for (int c = threadIdx.x; c < act_dim; c += blockDim.x)
      ract[idx * act_dim + c] = act[c];
    if (threadIdx.x == 0) {
      rrew[idx] = rew;
      rdone[idx] = done;
    }
  }
  const int64_t chunk = (frame_n + nslice - 1) / nslice;
  const int64_t lo = (int64_t)slice * chunk;
  const int64_t hi = min(frame_n, lo + chunk);
  if constexpr (QUANT) {
    uint8_t* f0 = (uint8_t*)rframe + idx * frame_n;
    uint8_t* f1 = (uint8_t*)rnframe + idx * frame_n;
    for (int64_t c = lo + threadIdx.x; c < hi; c += blockDim.x) {
      // matches the host encode: round((clamp(v,-1,1)+1)*127.5)
      float v0 = fminf(fmaxf(frame[c], -1.f), 1.f);
      float v1 = fminf(fmaxf(nframe[c], -1.f), 1.f);
      f0[c] = (uint8_t)lrintf((v0 + 1.f) * 127.5f);
      f1[c] = (uint8_t)lrintf((v1 + 1.f) * 127.5f);
    }
  } else {
    float* f0 = (float*)rframe + idx * frame_n;
    float* f1 = (float*)rnframe + idx * frame_n;
    for (int64_t c = lo + threadIdx.x; c < hi; c += blockDim.x) {
      f0[c] = frame[c];
      f1[c] = nframe[c];
    }
  }
}

void visual_store_into(torch::Tensor feat, torch::Tensor frame,
                       torch::Tensor nfeat, torch::Tensor nframe,
                       torch::Tensor act, double rew, double done,
                       torch::Tensor rfeat, torch::Tensor rframe,
                       torch::Tensor rnfeat, torch::Tensor rnframe,
                       torch::Tensor ract, torch::Tensor rrew,
                       torch::Tensor rdone, int64_t idx) {
  CHECK_IN(feat); CHECK_IN(rfeat);
  const int feat_dim = (int)feat.numel();
  const int act_dim = (int)act.numel();
  const int64_t frame_n = frame.numel();
  const bool quant = rframe.scalar_type() == torch::kUInt8;
  const int slices = (int)std::min<int64_t>(8, (frame_n + 4095) / 4096);
  auto s = cur_stream();
  auto L = [&](auto q) {
    hipLaunchKernelGGL((visual_store_kernel<decltype(q)::value>),
                       dim3(std::max(1, slices)), dim3(256), 0, s,
                       feat.data_ptr<float>(), frame.data_ptr<float>(),
                       nfeat.data_ptr<float>(), nframe.data_ptr<float>(),
                       act.data_ptr<float>(), (float)rew, (float)done,
                       rfeat.data_ptr<float>(), rframe.data_ptr(),
                       rnfeat.data_ptr<float>(), rnframe.data_ptr(),
                       ract.data_ptr<float>(), rrew.data_ptr<float>(),
                       rdone.data_ptr<float>(), idx, feat_dim, frame_n,
                       act_dim);
  };
  if (quant) L(std::true_type{});
  else L(std::false_type{});
}

void visual_sample_into(torch::Tensor feat, torch::Tensor frames,
                        torch::Tensor nfeat, torch::Tensor nframes,
                        torch::Tensor act, torch::Tensor rew,
                        torch::Tensor done, torch::Tensor size_dev,
                        torch::Tensor ctr, int64_t seed,
                        torch::Tensor of, torch::Tensor ofr,
                        torch::Tensor onf, torch::Tensor onfr,
                        torch::Tensor oa, torch::Tensor orew,
                        torch::Tensor od) {
  CHECK_IN(feat); CHECK_IN(of); CHECK_IN(ofr);
  const int B = (int)of.size(0);
  const int feat_dim = (int)feat.size(1);
  const int act_dim = (int)act.size(1);
  const int64_t frame_n = frames.numel() / frames.size(0);
  const bool quant = frames.scalar_type() == torch::kUInt8;
  auto s = cur_stream();
  hipLaunchKernelGGL(bump_counter_kernel, dim3(1), dim3(1), 0, s,
                     ctr.data_ptr<int64_t>());
  const int slices = (int)std::min<int64_t>(8, (frame_n + 4095) / 4096);
  dim3 grid(B, std::max(1, slices));
  if (quant)
    hipLaunchKernelGGL((visual_gather_kernel<true>), grid, dim3(256),
                       0, s, feat.data_ptr<float>(), frames.data_ptr(),
                       nfeat.data_ptr<float>(), nframes.data_ptr(),
                       act.data_ptr<float>(), rew.data_ptr<float>(),
                       done.data_ptr<float>(),
                       size_dev.data_ptr<int64_t>(),
                       ctr.data_ptr<int64_t>(), (uint64_t)seed,
                       of.data_ptr<float>(), ofr.data_ptr<float>(),
                       onf.data_ptr<float>(), onfr.data_ptr<float>(),
                       oa.data_ptr<float>(), orew.data_ptr<float>(),
                       od.data_ptr<float>(), feat_dim, frame_n, act_dim);
  else
    hipLaunchKernelGGL((visual_gather_kernel<false>), grid, dim3(256),
                       0, s, feat.data_ptr<float>(), frames.data_ptr(),
                       nfeat.data_ptr<float>(), nframes.data_ptr(),
                       act.data_ptr<float>(), rew.data_ptr<float>(),
                       done.data_ptr<float>(),
                       size_dev.data_ptr<int64_t>(),
                       ctr.data_ptr<int64_t>(), (uint64_t)seed,
                       of.data_ptr<float>(), ofr.data_ptr<float>(),
                       onf.data_ptr<float>(), onfr.data_ptr<float>(),
                       oa.data_ptr<float>(), orew.data_ptr<float>(),
                       od.data_ptr<float>(), feat_dim, frame_n, act_dim);
}

void bump_counter(torch::Tensor ctr) {
  hipLaunchKernelGGL(bump_counter_kernel, dim3(1), dim3(1), 0, cur_stream(),
                     ctr.data_ptr<int64_t>());
}

void philox_randn_(torch::Tensor out, torch::Tensor ctr, int64_t seed) {
  CHECK_IN(out);
  int64_t n = out.numel();
  auto s = cur_stream();
  hipLaunchKernelGGL(bump_counter_kernel, dim3(1), dim3(1), 0, s,
                     ctr.data_ptr<int64_t>());
  int64_t n4 = (n + 3) / 4;
  int blocks = (int)std::min<int64_t>((n4 + 255) / 256, 2048);
  hipLaunchKernelGGL(philox_randn_kernel, dim3(blocks), dim3(256), 0, s,
                     out.data_ptr<float>(), n, ctr.data_ptr<int64_t>(),
                     (uint64_t)seed);
}

std::vector<torch::Tensor> replay_sample(torch::Tensor state, torch::Tensor act,
                                         torch::Tensor rew,
                                         torch::Tensor nstate,
                                         torch::Tensor done,
                                         torch::Tensor size_dev,
                                         torch::Tensor ctr, int64_t seed,
                                         int64_t batch) {
  auto os = torch::empty({batch, state.size(1)}, state.options());
  auto oa = torch::empty({batch, act.size(1)}, state.options());
  auto orew = torch::empty({batch}, state.options());
  auto ons = torch::empty({batch, state.size(1)}, state.options());
  auto od = torch::empty({batch}, state.options());
  replay_sample_into(state, act, rew, nstate, done, size_dev, ctr, seed,
                     os, oa, orew, ons, od);
  return {os, oa, orew, ons, od};
}

}  // namespace

// shared compute-mode flag for the fused-engine / conv TUs
namespace fused { bool* g_bf16_flag = nullptr; }
void register_fused(pybind11::module_& m);
void register_conv(pybind11::module_& m);
void set_conv_bf16_flag(bool* p);
void register_mlpf(pybind11::module_& m);
void set_mlpf_bf16_flag(bool* p);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  fused::g_bf16_flag = &g_bf16;
  set_conv_bf16_flag(&g_bf16);
  set_mlpf_bf16_flag(&g_bf16);
  register_fused(m);
  register_conv(m);
  register_mlpf(m);
  m.def("set_compute_bf16", &set_compute_bf16,
        "Switch GEMM kernels to bf16 MFMA inputs (fp32 accumulate)");
  m.def("get_compute_bf16", &get_compute_bf16);
  m.def("linear_fwd", &linear_fwd, "y = x @ w^T + b (+relu), MFMA");
  m.def("linear_bwd", &linear_bwd, "dx, dw, db");
  m.def("tanh_gauss_fwd", &tanh_gauss_fwd);
  m.def("tanh_gauss_bwd", &tanh_gauss_bwd);
  m.def("sac_q_loss_fwd", &sac_q_loss_fwd);
  m.def("sac_pi_loss_fwd", &sac_pi_loss_fwd);
  m.def("polyak_", &polyak_);
  m.def("adam_step_", &adam_step_);
  m.def("replay_sample", &replay_sample);
  m.def("replay_sample_into", &replay_sample_into);
  m.def("visual_sample_into", &visual_sample_into);
  m.def("visual_store_into", &visual_store_into);
  m.def("philox_randn_", &philox_randn_);
  m.def("bump_counter", &bump_counter);
}
