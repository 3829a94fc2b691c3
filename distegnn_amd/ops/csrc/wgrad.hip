// Split-K weight-gradient kernel: dW = g^T @ x for tall-skinny activations.
//
// Shapes: g [M, 64] bf16 (layer-output grad), x [M, I<=208] bf16
// (activation), M ~ 10^5..10^6. hipBLASLt's heuristic picks a non-split-K
// kernel here (6 workgroups on 256 CUs, 2.9 ms measured); torch's bmm
// split-K workaround costs 2.6 ms of HOST time per call re-running the
// batched-GEMM heuristic. This kernel owns the shape: each block reduces a
// row chunk with mfma_f32_16x16x32_bf16 over LDS-transposed sub-tiles and
// writes an fp32 partial; the [nchunk, 64, I] partials are summed by a
// tiny torch reduction.

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int O_DIM = 64;       // rows of dW (layer width) — fixed
constexpr int THREADS = 256;    // 4 waves
// E_STEP: MFMA K per staging round; T_STRIDE = E_STEP + 2 LDS e-stride —
// (T_STRIDE/2) odd makes the 16 b64 readers of a fragment hit distinct
// banks. Small-NTW shapes (I=64) are staging-bound and do better with the
// shorter round; wide shapes (I=144+) amortize staging over 3x the MFMAs.

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

// NTW = n-tiles (of 16 cols of x) per wave; ITILES total = waves used * NTW
// mapping: wave w covers n-tiles [w*NTW, w*NTW+NTW)
template <int NTW, int E_STEP, int T_STRIDE>
__global__ __launch_bounds__(THREADS, NTW <= 2 ? 4 : 3) void wgrad_splitk(
    const bf16* __restrict__ g,  // [M, 64]
    const bf16* __restrict__ x,  // [M, I]
    float* __restrict__ part,    // [nchunk, 64, IP] (IP = 16*ceil(I/16))
    long m, int i_dim, int ip, long chunk) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // LDS: gT [64][T_STRIDE] bf16, xT [ip<=208][T_STRIDE] bf16
  __bf16* gT = reinterpret_cast<__bf16*>(smem);
  __bf16* xT = gT + O_DIM * T_STRIDE;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int itiles = ip / 16;

  long c0 = (long)blockIdx.x * chunk;
  long c1 = c0 + chunk < m ? c0 + chunk : m;

  f32x4 acc[4][NTW] = {};

  // Register double-buffering (round 2): the next round's global loads of
  // g/x are issued DURING the current round's MFMA phase (they have no LDS
  // dependency), hiding the global latency the old stage->barrier->mfma
  // sequence exposed every E_STEP rows. Per-thread prefetch registers are
  // sized by the template's column capacity (ip <= NTW*64).
  constexpr int GITER = (E_STEP * (O_DIM / 8) + THREADS - 1) / THREADS;
  constexpr int XITER = (E_STEP * (NTW * 64 / 8) + THREADS - 1) / THREADS;
  bf16x8 pg[GITER];
  bf16x8 px[XITER];

  auto prefetch = [&](long e0) {
    int ne = (int)((c1 - e0 < E_STEP) ? (c1 - e0) : (long)E_STEP);
#pragma unroll
    for (int it = 0; it < GITER; ++it) {
      int idx = tid + it * THREADS;
      int e = idx / (O_DIM / 8);
      int o8 = (idx % (O_DIM / 8)) * 8;
      bf16x8 v = {};
      if (idx < E_STEP * (O_DIM / 8) && e < ne)
        v = *reinterpret_cast<const bf16x8*>(g + (e0 + e) * O_DIM + o8);
      pg[it] = v;
    }
#pragma unroll
    for (int it = 0; it < XITER; ++it) {
      int idx = tid + it * THREADS;
      int e = idx / (ip / 8);
      int i8 = (idx % (ip / 8)) * 8;
      bf16x8 v = {};
      if (idx < E_STEP * (ip / 8) && e < ne) {
        if (i8 + 8 <= i_dim) {
          v = *reinterpret_cast<const bf16x8*>(x + (e0 + e) * i_dim + i8);
        } else {
#pragma unroll
          for (int u = 0; u < 8; ++u) {
            int i = i8 + u;
            v[u] = i < i_dim ? ((const __bf16*)x)[(e0 + e) * i_dim + i]
                             : (__bf16)0.f;
          }
        }
      }
      px[it] = v;
    }
  };

  prefetch(c0);
  for (long e0 = c0; e0 < c1; e0 += E_STEP) {
    __syncthreads();
    // LDS-transpose the prefetched round: gT[o][e], xT[i][e]
#pragma unroll
    for (int it = 0; it < GITER; ++it) {
      int idx = tid + it * THREADS;
      if (idx >= E_STEP * (O_DIM / 8)) break;
      int e = idx / (O_DIM / 8);
      int o8 = (idx % (O_DIM / 8)) * 8;
      bf16x8 v = pg[it];
#pragma unroll
      for (int u = 0; u < 8; ++u) gT[(o8 + u) * T_STRIDE + e] = v[u];
    }
#pragma unroll
    for (int it = 0; it < XITER; ++it) {
      int idx = tid + it * THREADS;
      if (idx >= E_STEP * (ip / 8)) break;
      int e = idx / (ip / 8);
      int i8 = (idx % (ip / 8)) * 8;
      bf16x8 v = px[it];
#pragma unroll
      for (int u = 0; u < 8; ++u) xT[(i8 + u) * T_STRIDE + e] = v[u];
    }
    __syncthreads();
    if (e0 + E_STEP < c1) prefetch(e0 + E_STEP);
    // D[o][i] += gT[o][e] * xT[i][e] — A rows = o, B cols = i, K = e
#pragma unroll
    for (int kk = 0; kk < E_STEP / 32; ++kk) {
      int kb = kk * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int nt = 0; nt < NTW; ++nt) {
        int icol = (wave * NTW + nt) * 16 + (lane & 15);
        if (wave * NTW + nt >= itiles) break;
        bf16x8 b = *reinterpret_cast<const bf16x8*>(&xT[icol * T_STRIDE + kb]);
#pragma unroll
        for (int mt = 0; mt < 4; ++mt) {
          bf16x8 a = *reinterpret_cast<const bf16x8*>(
              &gT[(mt * 16 + (lane & 15)) * T_STRIDE + kb]);
          acc[mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, b, acc[mt][nt], 0, 0, 0);
        }
      }
    }
  }

  // write partial [64][ip]: C layout col=l&15(+16*(w*NTW+nt)), row=(l>>4)*4+r
  float* out = part + (long)blockIdx.x * O_DIM * ip;
#pragma unroll
  for (int nt = 0; nt < NTW; ++nt) {
    if (wave * NTW + nt >= itiles) break;
    int i = (wave * NTW + nt) * 16 + (lane & 15);
#pragma unroll
    for (int mt = 0; mt < 4; ++mt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int o = mt * 16 + (lane >> 4) * 4 + r;
        out[o * ip + i] = acc[mt][nt][r];
      }
    }
  }
}

// Row-streaming partial combine: each block owns (row-chunk, col-tile) and
// sums its rows with coalesced reads; a second pass folds the row-chunk
// partials. Fixed tree order -> deterministic. aten's sum(0) walks the
// [nchunk, 64*ip] array column-major (~700 GB/s for this shape).
__global__ void colsum_rows(const float* __restrict__ src,
                            float* __restrict__ dst, long rows, long cols,
                            long rows_per_chunk) {
  long nchunks = (rows + rows_per_chunk - 1) / rows_per_chunk;
  long ctiles = (cols + 255) / 256;
  for (long b = blockIdx.x; b < nchunks * ctiles; b += gridDim.x) {
    long rc = b / ctiles;
    long c = (b - rc * ctiles) * 256 + threadIdx.x;
    if (c >= cols) continue;
    long r0 = rc * rows_per_chunk;
    long r1 = r0 + rows_per_chunk < rows ? r0 + rows_per_chunk : rows;
    float acc = 0.f;
    for (long r = r0; r < r1; ++r) acc += src[r * cols + c];
    dst[rc * cols + c] = acc;
  }
}

}  // namespace

torch::Tensor wgrad_splitk_launch(torch::Tensor g, torch::Tensor x) {
  TORCH_CHECK(g.is_cuda() && g.scalar_type() == torch::kBFloat16,
              "g must be CUDA bf16");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "x must be bf16");
  TORCH_CHECK(g.size(1) == O_DIM, "wgrad kernel requires out width 64");
  auto gc = g.contiguous();
  auto xc = x.contiguous();
  long m = gc.size(0);
  int i_dim = (int)xc.size(1);
  TORCH_CHECK(i_dim <= 208, "wgrad kernel supports I<=208");
  int ip = ((i_dim + 15) / 16) * 16;
  long chunk = ((m + 1023) / 1024 + 31) / 32 * 32;  // ~1024 blocks
  if (chunk < 64) chunk = 64;
  long nchunk = (m + chunk - 1) / chunk;
  auto part = torch::empty({nchunk, (long)O_DIM, (long)ip},
                           g.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  int t_stride = (((i_dim + 15) / 16 + 3) / 4 >= 3) ? 66 : 40;
  int smem = (O_DIM + ip) * t_stride * 2;
  int itiles = ip / 16;
  int ntw = (itiles + 3) / 4;
  const bf16* gp = reinterpret_cast<const bf16*>(gc.data_ptr());
  const bf16* xp = reinterpret_cast<const bf16*>(xc.data_ptr());
  float* pp = part.data_ptr<float>();
  switch (ntw) {
    case 1:
      wgrad_splitk<1, 32, 40><<<(int)nchunk, THREADS, smem, stream>>>(
          gp, xp, pp, m, i_dim, ip, chunk);
      break;
    case 2:
      wgrad_splitk<2, 32, 40><<<(int)nchunk, THREADS, smem, stream>>>(
          gp, xp, pp, m, i_dim, ip, chunk);
      break;
    case 3:
      wgrad_splitk<3, 64, 66><<<(int)nchunk, THREADS, smem, stream>>>(
          gp, xp, pp, m, i_dim, ip, chunk);
      break;
    case 4:
      wgrad_splitk<4, 64, 66><<<(int)nchunk, THREADS, smem, stream>>>(
          gp, xp, pp, m, i_dim, ip, chunk);
      break;
    default:
      TORCH_CHECK(false, "unsupported I for wgrad kernel");
  }
  // two-pass row-streaming combine of the [nchunk, 64*ip] partials
  long cols = (long)O_DIM * ip;
  auto dw = torch::empty({(long)O_DIM, (long)ip},
                         g.options().dtype(torch::kFloat));
  if (nchunk <= 16) {
    colsum_rows<<<num_blocks(((cols + 255) / 256) * 256, 256), 256, 0,
                  stream>>>(pp, dw.data_ptr<float>(), nchunk, cols, nchunk);
  } else {
    long rpc = 16;
    long nmid = (nchunk + rpc - 1) / rpc;
    auto mid = torch::empty({nmid, cols}, g.options().dtype(torch::kFloat));
    colsum_rows<<<num_blocks(nmid * ((cols + 255) / 256) * 256, 256), 256, 0,
                  stream>>>(pp, mid.data_ptr<float>(), nchunk, cols, rpc);
    colsum_rows<<<num_blocks(((cols + 255) / 256) * 256, 256), 256, 0,
                  stream>>>(mid.data_ptr<float>(), dw.data_ptr<float>(),
                            nmid, cols, nmid);
  }
  return dw.narrow(1, 0, i_dim);
}
