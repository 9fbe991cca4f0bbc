// Whole-MLP fused forward (gfx950 / CDNA4).
//
// One kernel launch runs an entire MLP stack for a 64-row block: the
// input rows stage into LDS once, every layer's GEMM reads its
// predecessor's output straight from LDS (activations never round-trip
// through HBM between layers), each wave owns a 64-column output strip
// with PRIVATE weight staging (no intra-layer barriers — one barrier per
// layer), MFMA bf16 (fp32-exact mode for parity tests).  blockIdx.z
// selects the problem (both twin critics in one launch); a layer may
// split its output rows across two weight tensors (the policy's
// mu/log_std dual head).  Hidden activations are also written to global
// buffers for the backward pass.
//
// Replaces 3 layer-GEMM launches per MLP evaluation with 1; the fused
// SAC update runs 4 of these per update (stacked actor, target critic
// pair, critic pair x2 phases).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <vector>

namespace mlpf {

#define DEVINL __device__ __forceinline__

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

constexpr int MAXL = 5;
constexpr int BK = 64;          // K-chunk for weight staging

struct Args {
  const float* x; int64_t x_off; int ldx;
  int M, K0, n_layers, relu_mask;
  int width[MAXL];
  int split[MAXL];              // output rows >= split come from whi/bhi
  const float* w[2][MAXL];
  const float* whi[2][MAXL];
  const float* bias[2][MAXL];
  const float* bhi[2][MAXL];
  float* act[2][MAXL];          // [M, width[L]] fp32 (nullable)
  int ldsx;                     // LDS elements per X row (aligned+pad)
  int ldsw;                     // LDS elements per W strip row (BK+pad)
};

template <bool BF16> struct Elt;
template <> struct Elt<true> { using T = __bf16; };
template <> struct Elt<false> { using T = float; };

template <bool BF16>
DEVINL const float* wrow_ptr(const Args& a, int z, int L, int row, int* r) {
  if (row >= a.split[L]) { *r = row - a.split[L]; return a.whi[z][L]; }
  *r = row;
  return a.w[z][L];
}

template <bool BF16>
__global__ __launch_bounds__(256)
void mlp_fwd_kernel(Args a) {
  using T = typename Elt<BF16>::T;
  extern __shared__ __attribute__((aligned(16))) char dyn[];
  const int z = blockIdx.z;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int bm0 = blockIdx.x * 64;

  T* XA = (T*)dyn;
  T* XB = XA + 64 * a.ldsx;
  T* WS = XB + 64 * a.ldsx + wid * 64 * a.ldsw;   // this wave's strip

  // ---- stage input rows (zero-padded to the BK-aligned width) -------
  {
    const int row = tid & 63;
    const int grow = bm0 + row;
    const int kpad = ((a.K0 + BK - 1) / BK) * BK;
    for (int c = tid >> 6; c < kpad; c += 4) {
      float v = 0.f;
      if (grow < a.M && c < a.K0)
        v = a.x[a.x_off + (int64_t)grow * a.ldx + c];
      XA[row * a.ldsx + c] = (T)v;
    }
  }
  __syncthreads();

  T* cur = XA;
  T* nxt = XB;
  int K = a.K0;

  for (int L = 0; L < a.n_layers; ++L) {
    const int H = a.width[L];
    const bool relu = (a.relu_mask >> L) & 1;
    const int wc0 = wid * 64;           // this wave's output-column base
    const int hpad = ((H + BK - 1) / BK) * BK;

    if (wc0 < H) {
      f32x4 acc[4][4] = {};
      const int kpad = ((K + BK - 1) / BK) * BK;
      const int row = wc0 + lane;
      int rr = 0;
      const float* wsrc = row < H
          ? wrow_ptr<BF16>(a, z, L, row, &rr) : nullptr;
      const int Kc = K;
      float vw[BK];
      auto load_w = [&](int k0) {
        if (wsrc && ((Kc & 3) == 0) && (k0 + BK <= Kc)) {
          const float4* s = (const float4*)(wsrc + (int64_t)rr * Kc + k0);
#pragma unroll
          for (int q = 0; q < BK / 4; ++q) {
            float4 f = s[q];
            vw[q*4+0]=f.x; vw[q*4+1]=f.y; vw[q*4+2]=f.z; vw[q*4+3]=f.w;
          }
        } else {
#pragma unroll
          for (int c = 0; c < BK; ++c)
            vw[c] = (wsrc && k0 + c < Kc)
                        ? wsrc[(int64_t)rr * Kc + k0 + c] : 0.f;
        }
      };
      load_w(0);
      for (int k0 = 0; k0 < kpad; k0 += BK) {
        // write the prefetched W chunk to this wave's LDS strip
        if constexpr (BF16) {
          union { __bf16 h[BK]; uint4 u[BK / 8]; } pk;
#pragma unroll
          for (int c = 0; c < BK; ++c) pk.h[c] = (__bf16)vw[c];
          uint4* dst = (uint4*)&WS[lane * a.ldsw];
#pragma unroll
          for (int q = 0; q < BK / 8; ++q) dst[q] = pk.u[q];
        } else {
          float* dst = (float*)&WS[lane * a.ldsw];
#pragma unroll
          for (int c = 0; c < BK; ++c) dst[c] = vw[c];
        }
        // prefetch the next chunk while the MFMAs run (T14)
        if (k0 + BK < kpad) load_w(k0 + BK);
        // wave-private strip: no barrier needed; MFMA over the chunk
        if constexpr (BF16) {
          const int arow = lane & 15;
          const int ak0 = (lane >> 4) * 8;
#pragma unroll
          for (int kk = 0; kk < BK; kk += 32) {
#pragma unroll
            for (int mi = 0; mi < 4; ++mi) {
              bf16x8 av = *(const bf16x8*)&cur[(mi * 16 + arow) * a.ldsx
                                               + k0 + kk + ak0];
#pragma unroll
              for (int ni = 0; ni < 4; ++ni) {
                bf16x8 bv = *(const bf16x8*)&WS[(ni * 16 + arow) * a.ldsw
                                                + kk + ak0];
                acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    av, bv, acc[mi][ni], 0, 0, 0);
              }
            }
          }
        } else {
          const int arow = lane & 15;
          const int akl = lane >> 4;
#pragma unroll
          for (int kk = 0; kk < BK; kk += 4) {
#pragma unroll
            for (int mi = 0; mi < 4; ++mi) {
              float av = ((const float*)cur)[(mi * 16 + arow) * a.ldsx
                                             + k0 + kk + akl];
#pragma unroll
              for (int ni = 0; ni < 4; ++ni) {
                float bv = ((const float*)WS)[(ni * 16 + arow) * a.ldsw
                                              + kk + akl];
                acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                    av, bv, acc[mi][ni], 0, 0, 0);
              }
            }
          }
        }
      }
      // epilogue: bias (+relu), write LDS (next input) + global act
      const int crow = (lane >> 4) * 4;
      const int ccol = lane & 15;
      float* actp = a.act[z][L];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int row = mi * 16 + crow + r;
            const int col = wc0 + ni * 16 + ccol;
            if (col < H) {
              int br;
              const float* bsrc = a.bias[z][L];
              int bcol = col;
              if (col >= a.split[L]) { bsrc = a.bhi[z][L];
                                       bcol = col - a.split[L]; }
              float v = acc[mi][ni][r] + (bsrc ? bsrc[bcol] : 0.f);
              if (relu) v = fmaxf(v, 0.f);
              nxt[row * a.ldsx + col] = (T)v;
              if (actp && bm0 + row < a.M)
                actp[(int64_t)(bm0 + row) * H + col] = v;
              (void)br;
            }
          }
    }
    // zero the BK pad of the new activation block
    {
      const int row = tid & 63;
      for (int c = H + (tid >> 6); c < hpad; c += 4)
        nxt[row * a.ldsx + c] = (T)0.f;
    }
    __syncthreads();
    T* tmp = cur; cur = nxt; nxt = tmp;
    K = H;
  }
}

inline hipStream_t stream() { return c10::hip::getCurrentHIPStream().stream(); }

extern bool* g_bf16_flag3;

// layers: per layer a dict-free tuple list from Python:
//   (w_lo[z=0], w_lo[z=1] | None, w_hi|None, b_lo.., b_hi.., act[z]...)
// Assembled host-side into Args.
void mlp_fwd_fused(
    torch::Tensor x, int64_t x_off, int64_t ldx, int64_t M, int64_t K0,
    std::vector<std::vector<torch::Tensor>> w,        // [z][L]
    std::vector<std::vector<torch::Tensor>> whi,      // [z][L] (may be empty per L -> undefined tensor)
    std::vector<std::vector<torch::Tensor>> bias,
    std::vector<std::vector<torch::Tensor>> bhi,
    std::vector<std::vector<torch::Tensor>> act,      // [z][L]
    std::vector<int64_t> widths, std::vector<int64_t> splits,
    int64_t relu_mask) {
  const int nz = (int)w.size();
  const int nL = (int)widths.size();
  TORCH_CHECK(nz >= 1 && nz <= 2 && nL <= MAXL);
  Args a{};
  a.x = x.data_ptr<float>();
  a.x_off = x_off;
  a.ldx = (int)ldx;
  a.M = (int)M;
  a.K0 = (int)K0;
  a.n_layers = nL;
  a.relu_mask = (int)relu_mask;
  int maxdim = (int)K0;
  for (int L = 0; L < nL; ++L) {
    a.width[L] = (int)widths[L];
    a.split[L] = (int)splits[L];
    maxdim = std::max(maxdim, a.width[L]);
    for (int z = 0; z < nz; ++z) {
      a.w[z][L] = w[z][L].data_ptr<float>();
      auto opt = [](const torch::Tensor& t) -> const float* {
        return (t.defined() && t.numel() > 0) ? t.data_ptr<float>()
                                              : nullptr;
      };
      a.whi[z][L] = opt(whi[z][L]);
      a.bias[z][L] = opt(bias[z][L]);
      a.bhi[z][L] = opt(bhi[z][L]);
      a.act[z][L] = const_cast<float*>(opt(act[z][L]));
    }
  }
  const bool bf16 = *g_bf16_flag3;
  const int elt = bf16 ? 2 : 4;
  const int pad = bf16 ? 8 : 1;
  a.ldsx = ((maxdim + BK - 1) / BK) * BK + pad;
  a.ldsw = BK + pad;
  const size_t lds = (size_t)(2 * 64 * a.ldsx + 4 * 64 * a.ldsw) * elt;
  TORCH_CHECK(lds <= 160 * 1024, "fused MLP forward: LDS budget exceeded");
  dim3 grid((a.M + 63) / 64, 1, nz);
  if (bf16)
    hipLaunchKernelGGL((mlp_fwd_kernel<true>), grid, dim3(256), lds,
                       stream(), a);
  else
    hipLaunchKernelGGL((mlp_fwd_kernel<false>), grid, dim3(256), lds,
                       stream(), a);
}

// host-side feasibility probe (mirrors the LDS computation above)
bool mlp_fwd_fits(int64_t K0, std::vector<int64_t> widths, bool bf16) {
  int maxdim = (int)K0;
  for (auto wd : widths) maxdim = std::max(maxdim, (int)wd);
  const int elt = bf16 ? 2 : 4;
  const int pad = bf16 ? 8 : 1;
  int ldsx = ((maxdim + BK - 1) / BK) * BK + pad;
  size_t lds = (size_t)(2 * 64 * ldsx + 4 * 64 * (BK + pad)) * elt;
  return lds <= 160 * 1024 && (int)widths.size() <= MAXL;
}

}  // namespace mlpf

namespace mlpf { bool* g_bf16_flag3 = nullptr; }

void set_mlpf_bf16_flag(bool* p) { mlpf::g_bf16_flag3 = p; }

void register_mlpf(pybind11::module_& m) {
  m.def("mlp_fwd_fused", &mlpf::mlp_fwd_fused);
  m.def("mlp_fwd_fits", &mlpf::mlp_fwd_fits);
}
