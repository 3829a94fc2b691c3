// Fused SchNet continuous-filter convolution message kernel (gfx950).
//
// Computes, per edge e = (row, col):
//   gauss_k = exp(coeff * (d_e - offset_k)^2)            k < G   (smearing)
//   z1      = ssp(gauss @ W1f^T + b1)                    [F]     (filter MLP)
//   w       = z1 @ W2f^T + b2                            [F]
//   msg_e   = xw1[col_e] * w * 0.5*(cos(d_e pi/cutoff)+1)
//
// replacing the eager chain GaussianSmearing -> Linear -> ssp -> Linear ->
// cutoff-mul -> gather -> mul (reference SchNet.py:304-354; our
// models/schnet.py CFConv) and its [M,G]/[M,F] materializations with one
// kernel producing the per-edge messages. The caller aggregates with the
// CSR segment-sum kernels and applies lin2.
//
// Tiling follows fused_edge.hip: one 4-wave workgroup owns a 64-edge tile,
// MFMA 16x16x32 bf16 with fp32 accumulation, B-fragments read K-contiguous
// from L2-resident padded weights, LDS staging regions time-shared
// (gauss -> free; z1 -> xw1 gather). Templated over F in {64, 128}
// (FastSchNet H=64 headline / SchNet 128 default); G <= 64 zero-padded.
// ssp(x) = softplus(x) - log 2 (reference ShiftedSoftplus).

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int TILE = 64;
constexpr int THREADS = 256;
constexpr int GPAD = 64;       // padded gaussian count
constexpr int G_STRIDE = 72;

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ float ssp_(float x) {
  // log(1 + e^x) - log 2, overflow-safe
  float m = fmaxf(x, 0.f);
  return m + __logf(__expf(x - m) + __expf(-m)) - 0.6931471805599453f;
}

__device__ __forceinline__ bf16x8 lds8(const char* smem, int off) {
  return *reinterpret_cast<const bf16x8*>(smem + off);
}
__device__ __forceinline__ bf16x8 g8(const bf16* p) {
  return *reinterpret_cast<const bf16x8*>(p);
}
__device__ __forceinline__ const bf16* opaque(const bf16* p) {
  asm volatile("" : "+v"(p));
  return p;
}

template <int F>
__global__ __launch_bounds__(THREADS) void cfconv_fwd(
    const bf16* __restrict__ xw1,      // [N, F] = lin1(x)
    const float* __restrict__ dist,    // [M]
    const long* __restrict__ row, const long* __restrict__ col,
    const bf16* __restrict__ w1f,      // [F][GPAD] padded filter L1
    const float* __restrict__ b1,      // [F]
    const bf16* __restrict__ w2f,      // [F][F]
    const float* __restrict__ b2,      // [F]
    const float* __restrict__ offsets,  // [GPAD] (pad: +inf -> gauss 0)
    bf16* __restrict__ msg_out,        // [M, F]
    long m, float coeff, float inv_cutoff_pi, int g_real) {
  constexpr int F_STRIDE = F + 8;
  constexpr int NT = F / 16;           // n-tiles across all 4 waves' rows
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // regions: A gauss [TILE][G_STRIDE], B z1/xj [TILE][F_STRIDE],
  //          C w [TILE][F_STRIDE], scal [TILE] f32 (cutoff factor),
  //          bias [2F] f32, cols [TILE] i32
  const int A = 0;
  const int B = A + TILE * G_STRIDE * 2;
  const int C = B + TILE * F_STRIDE * 2;
  const int SC = C + TILE * F_STRIDE * 2;
  const int BIAS = SC + TILE * 4;
  const int COLS = BIAS + 2 * F * 4;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  float* biases = reinterpret_cast<float*>(smem + BIAS);
  for (int i = tid; i < F; i += THREADS) {
    biases[i] = b1[i];
    biases[F + i] = b2[i];
  }

  for (long tile = blockIdx.x; tile * TILE < m; tile += gridDim.x) {
    long e0 = tile * TILE;
    int nedge = (int)((m - e0 < (long)TILE) ? (m - e0) : (long)TILE);
    __syncthreads();
    // ---- smearing + cutoff factor + col staging ----
    int* cls = reinterpret_cast<int*>(smem + COLS);
    for (int e = tid; e < TILE; e += THREADS) {
      float d = 0.f;
      if (e < nedge) {
        d = dist[e0 + e];
        cls[e] = (int)col[e0 + e];
      } else {
        cls[e] = 0;
      }
      reinterpret_cast<float*>(smem + SC)[e] =
          0.5f * (__cosf(d * inv_cutoff_pi) + 1.f);
      __bf16* grow = reinterpret_cast<__bf16*>(smem + A) + e * G_STRIDE;
      for (int k = 0; k < GPAD; ++k) {
        float dk = d - offsets[k];
        grow[k] = (__bf16)(k < g_real ? __expf(coeff * dk * dk) : 0.f);
      }
    }
    __syncthreads();
    // ---- z1 = ssp(gauss @ W1f^T + b1) ----
    {
      f32x4 acc[NT] = {};
#pragma unroll
      for (int kk = 0; kk < GPAD / 32; ++kk) {
        int k = kk * 32 + (lane >> 4) * 8;
        bf16x8 a = lds8(smem, A + ((wave * 16 + (lane & 15)) * G_STRIDE
                                   + k) * 2);
        const bf16* wp = opaque(w1f);
#pragma unroll
        for (int nt = 0; nt < NT; ++nt) {
          bf16x8 b = g8(wp + (nt * 16 + (lane & 15)) * GPAD + k);
          acc[nt] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0,
                                                      0);
        }
      }
      __bf16* z1 = reinterpret_cast<__bf16*>(smem + B);
#pragma unroll
      for (int nt = 0; nt < NT; ++nt) {
        int c = nt * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int e = wave * 16 + (lane >> 4) * 4 + r;
          z1[e * F_STRIDE + c] = (__bf16)ssp_(acc[nt][r] + biases[c]);
        }
      }
    }
    __syncthreads();
    // ---- w = z1 @ W2f^T + b2 ----
    {
      f32x4 acc[NT] = {};
#pragma unroll
      for (int kk = 0; kk < F / 32; ++kk) {
        int k = kk * 32 + (lane >> 4) * 8;
        bf16x8 a = lds8(smem, B + ((wave * 16 + (lane & 15)) * F_STRIDE
                                   + k) * 2);
        const bf16* wp = opaque(w2f);
#pragma unroll
        for (int nt = 0; nt < NT; ++nt) {
          bf16x8 b = g8(wp + (nt * 16 + (lane & 15)) * F + k);
          acc[nt] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0,
                                                      0);
        }
      }
      __bf16* w = reinterpret_cast<__bf16*>(smem + C);
#pragma unroll
      for (int nt = 0; nt < NT; ++nt) {
        int c = nt * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int e = wave * 16 + (lane >> 4) * 4 + r;
          w[e * F_STRIDE + c] = (__bf16)(acc[nt][r] + biases[F + c]);
        }
      }
    }
    __syncthreads();
    // ---- gather xw1[col] into B (z1 consumed), coalesced ----
    for (int idx = tid; idx < TILE * (F / 8); idx += THREADS) {
      int e = idx / (F / 8);
      int c8 = (idx % (F / 8)) * 8;
      bf16x8 v = {};
      if (e < nedge) v = g8(xw1 + (long)cls[e] * F + c8);
      *reinterpret_cast<bf16x8*>(smem + B + (e * F_STRIDE + c8) * 2) = v;
    }
    __syncthreads();
    // ---- msg = xj * w * cutoff ----
    for (int idx = tid; idx < TILE * (F / 8); idx += THREADS) {
      int e = idx / (F / 8);
      if (e >= nedge) continue;
      int c8 = (idx % (F / 8)) * 8;
      bf16x8 xj = lds8(smem, B + (e * F_STRIDE + c8) * 2);
      bf16x8 w = lds8(smem, C + (e * F_STRIDE + c8) * 2);
      float cf = reinterpret_cast<const float*>(smem + SC)[e];
      bf16x8 out;
#pragma unroll
      for (int u = 0; u < 8; ++u)
        out[u] = (__bf16)((float)xj[u] * (float)w[u] * cf);
      *reinterpret_cast<bf16x8*>(msg_out + (e0 + e) * F + c8) = out;
    }
  }
}

}  // namespace

torch::Tensor cfconv_forward(torch::Tensor xw1, torch::Tensor dist,
                             torch::Tensor row, torch::Tensor col,
                             torch::Tensor w1f, torch::Tensor b1,
                             torch::Tensor w2f, torch::Tensor b2,
                             torch::Tensor offsets, double coeff,
                             double cutoff) {
  TORCH_CHECK(xw1.is_cuda() && xw1.scalar_type() == torch::kBFloat16,
              "xw1 must be CUDA bf16");
  long f = xw1.size(1);
  TORCH_CHECK(f == 64 || f == 128, "fused cfconv supports F in {64,128}");
  long g = offsets.numel();
  TORCH_CHECK(g <= GPAD, "fused cfconv supports num_gaussians <= 64");
  long m = row.numel();
  auto msg = torch::empty({m, f}, xw1.options());
  if (m == 0) return msg;
  auto xc = xw1.contiguous();
  auto dc = dist.contiguous().to(torch::kFloat);
  auto w1c = w1f.contiguous();
  TORCH_CHECK(w1c.size(1) == GPAD && w1c.size(0) == f,
              "w1f must be [F][64] padded");
  auto w2c = w2f.contiguous();
  auto b1c = b1.contiguous().to(torch::kFloat);
  auto b2c = b2.contiguous().to(torch::kFloat);
  auto offp = torch::constant_pad_nd(
      offsets.contiguous().to(torch::kFloat), {0, GPAD - g});
  auto stream = at::hip::getCurrentHIPStream();
  long tiles = (m + TILE - 1) / TILE;
  int blocks = (int)std::min<long>(tiles, 16384);
  float inv_cutoff_pi = (float)(M_PI / cutoff);
#define LAUNCH(FF)                                                          \
  do {                                                                      \
    constexpr int F_STRIDE = FF + 8;                                        \
    int smem = TILE * G_STRIDE * 2 + 2 * TILE * F_STRIDE * 2 + TILE * 4 +   \
               2 * FF * 4 + TILE * 4;                                       \
    cfconv_fwd<FF><<<blocks, THREADS, smem, stream>>>(                      \
        reinterpret_cast<const bf16*>(xc.data_ptr()),                       \
        dc.data_ptr<float>(), row.contiguous().data_ptr<long>(),            \
        col.contiguous().data_ptr<long>(),                                  \
        reinterpret_cast<const bf16*>(w1c.data_ptr()),                      \
        b1c.data_ptr<float>(),                                              \
        reinterpret_cast<const bf16*>(w2c.data_ptr()),                      \
        b2c.data_ptr<float>(), offp.data_ptr<float>(),                      \
        reinterpret_cast<bf16*>(msg.data_ptr()), m, (float)coeff,           \
        inv_cutoff_pi, (int)g);                                             \
  } while (0)
  if (f == 64) LAUNCH(64);
  else LAUNCH(128);
#undef LAUNCH
  return msg;
}
