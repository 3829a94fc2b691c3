// Tall-skinny Linear kernel: y = act(x @ B + bias) for M ~ 10^5..10^6 rows.
//
// Covers the model's Linear forwards (B = W^T, W [O,K] torch layout) and
// data gradients (B = W, staged transposed), K,O <= 208. hipBLASLt's picks
// for these shapes run up to 12x below the bandwidth roofline (e.g.
// [565K,64] @ [64,134] dgrad: 467 us vs ~40 us) and cost ~30 us host per
// call; this kernel owns the shape class.
//
// Mapping: 4 waves/block, 64 rows/block; x rows staged to LDS
// (bank-padded), B staged once per block; per wave a 16-row x O MFMA tile
// (mfma_f32_16x16x32_bf16, fp32 accum), fused bias + activation epilogue,
// coalesced stores via LDS. NT (output 16-col tiles) is a compile-time
// template parameter.

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int TILE = 64;
constexpr int THREADS = 256;
constexpr int KMAX = 224;

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ float silu2(float x) {
  return x / (1.f + __expf(-x));
}

// act: 0 = none, 1 = silu
template <int NT, int ACT>
__global__ __launch_bounds__(THREADS) void tall_linear_kernel(
    const bf16* __restrict__ x,   // [M, K]
    const bf16* __restrict__ bmat,  // [O, K] (k-contiguous rows = B^T cols)
    const float* __restrict__ bias,  // [O] or nullptr
    bf16* __restrict__ y,         // [M, O]
    long m, int k_dim, int o_dim, int kp, int ks) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // LDS: x_tile [TILE][ks], b_tile [NT*16][ks], bias [O], y_tile [TILE][os]
  const int os = NT * 16 + 8;
  char* xs = smem;
  char* bs = xs + TILE * ks * 2;
  float* bvec = reinterpret_cast<float*>(bs + NT * 16 * ks * 2);
  __bf16* ys = reinterpret_cast<__bf16*>(
      reinterpret_cast<char*>(bvec) + ((o_dim + 3) / 4) * 4 * 4);
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  // stage B + bias once
  for (int idx = tid; idx < NT * 16 * kp / 8; idx += THREADS) {
    int r = idx / (kp / 8);
    int c8 = (idx % (kp / 8)) * 8;
    bf16x8 v = {};
    if (r < o_dim) {
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        int c = c8 + u;
        v[u] = (c < k_dim) ? ((const __bf16*)bmat)[r * k_dim + c]
                           : (__bf16)0.f;
      }
    }
    *reinterpret_cast<bf16x8*>(bs + r * ks * 2 + c8 * 2) = v;
  }
  for (int i = tid; i < o_dim; i += THREADS)
    bvec[i] = bias ? bias[i] : 0.f;

  for (long t = blockIdx.x; t * TILE < m; t += gridDim.x) {
    long r0 = t * TILE;
    int nrow = (int)((m - r0 < (long)TILE) ? (m - r0) : (long)TILE);
    __syncthreads();
    // stage x rows
    for (int idx = tid; idx < TILE * kp / 8; idx += THREADS) {
      int r = idx / (kp / 8);
      int c8 = (idx % (kp / 8)) * 8;
      bf16x8 v = {};
      if (r < nrow) {
        if (c8 + 8 <= k_dim) {
          v = *reinterpret_cast<const bf16x8*>(x + (r0 + r) * k_dim + c8);
        } else {
#pragma unroll
          for (int u = 0; u < 8; ++u) {
            int c = c8 + u;
            v[u] = (c < k_dim) ? ((const __bf16*)x)[(r0 + r) * k_dim + c]
                               : (__bf16)0.f;
          }
        }
      }
      *reinterpret_cast<bf16x8*>(xs + r * ks * 2 + c8 * 2) = v;
    }
    __syncthreads();

    f32x4 acc[NT] = {};
    int kb = (lane >> 4) * 8;
#pragma unroll
    for (int kk = 0; kk < KMAX / 32; ++kk) {
      if (kk * 32 >= kp) break;
      int k = kk * 32 + kb;
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          xs + ((wave * 16 + (lane & 15)) * ks + k) * 2);
#pragma unroll
      for (int nt = 0; nt < NT; ++nt) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            bs + ((nt * 16 + (lane & 15)) * ks + k) * 2);
        acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt],
                                                          0, 0, 0);
      }
    }
    // epilogue into LDS (C layout), then coalesced store
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) {
      int c = nt * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int e = wave * 16 + (lane >> 4) * 4 + r;
        float v = acc[nt][r] + (c < o_dim ? bvec[c] : 0.f);
        if (ACT == 1) v = silu2(v);
        ys[e * os + c] = (__bf16)v;
      }
    }
    __syncthreads();
    for (int idx = tid; idx < TILE * o_dim; idx += THREADS) {
      int r = idx / o_dim;
      int c = idx % o_dim;
      if (r < nrow) y[(r0 + r) * o_dim + c] = (bf16)ys[r * os + c];
    }
  }
}

}  // namespace

// bmat: [O, K] with K contiguous (for y = x @ bmat^T, torch Linear).
torch::Tensor tall_linear(torch::Tensor x, torch::Tensor bmat,
                          c10::optional<torch::Tensor> bias, int64_t act) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16,
              "x must be CUDA bf16");
  TORCH_CHECK(bmat.scalar_type() == torch::kBFloat16, "B must be bf16");
  auto xc = x.contiguous();
  auto bc = bmat.contiguous();
  long m = xc.size(0);
  int k_dim = (int)xc.size(1);
  int o_dim = (int)bc.size(0);
  TORCH_CHECK(bc.size(1) == k_dim, "shape mismatch");
  TORCH_CHECK(k_dim <= KMAX && o_dim <= KMAX, "K/O too large");
  auto y = torch::empty({m, (long)o_dim}, x.options());
  if (m == 0) return y;
  int kp = ((k_dim + 31) / 32) * 32;
  int ks = kp + 8;  // bank pad
  int nt = (o_dim + 15) / 16;
  TORCH_CHECK(nt <= 13, "O too large");
  int ntv = nt <= 9 ? nt : 13;  // instantiated template sizes
  int os = ntv * 16 + 8;
  int smem = TILE * ks * 2 + ntv * 16 * ks * 2 + ((o_dim + 3) / 4) * 4 * 4 +
             TILE * os * 2 + 64;
  torch::Tensor bias_t;
  const float* bias_p = nullptr;
  if (bias.has_value()) {
    bias_t = bias->contiguous().to(torch::kFloat);
    bias_p = bias_t.data_ptr<float>();
  }
  auto stream = at::hip::getCurrentHIPStream();
  long tiles = (m + TILE - 1) / TILE;
  int blocks = (int)std::min<long>(tiles, 8192);
  const bf16* xp = reinterpret_cast<const bf16*>(xc.data_ptr());
  const bf16* bp = reinterpret_cast<const bf16*>(bc.data_ptr());
  bf16* yp = reinterpret_cast<bf16*>(y.data_ptr());

#define LAUNCH(NTV)                                                         \
  do {                                                                      \
    if (act == 1)                                                           \
      tall_linear_kernel<NTV, 1><<<blocks, THREADS, smem, stream>>>(        \
          xp, bp, bias_p, yp, m, k_dim, o_dim, kp, ks);                     \
    else                                                                    \
      tall_linear_kernel<NTV, 0><<<blocks, THREADS, smem, stream>>>(        \
          xp, bp, bias_p, yp, m, k_dim, o_dim, kp, ks);                     \
  } while (0)

  switch (ntv) {
    case 1: LAUNCH(1); break;
    case 2: LAUNCH(2); break;
    case 3: LAUNCH(3); break;
    case 4: LAUNCH(4); break;
    case 5: LAUNCH(5); break;
    case 6: LAUNCH(6); break;
    case 7: LAUNCH(7); break;
    case 8: LAUNCH(8); break;
    case 9: LAUNCH(9); break;
    default: LAUNCH(13); break;
  }
#undef LAUNCH
  return y;
}
