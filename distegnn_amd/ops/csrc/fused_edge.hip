// Fused FastEGNN edge block — the north-star MFMA kernel (gfx950).
//
// Per edge e = (i=row[e], j=col[e]) this computes, in ONE kernel:
//   in_e   = [h_i | h_j | r_e | a_e]            (K = 2H+1+Ea = 131, H=64)
//   r_e    = ||x_i - x_j||^2, d_e = x_i - x_j   (optionally normalized)
//   t1     = SiLU(in_e  @ W1^T + b1)            [H]
//   msg    = SiLU(t1    @ W2^T + b2)            [H]   (edge_feat, phi_e)
//   s3     = SiLU(msg   @ W3^T + b3)            [H]   (phi_x hidden)
//   p_e    = s3 . w3                            scalar (phi_x head)
//   trans  = d_e * p_e                          [3]
// and writes msg [M,H] bf16 + trans [M,3] f32 (aggregated to nodes by the
// CSR segment-mean kernel). This replaces, per layer: 2 gathers, a [M,131]
// concat, 3 hipBLASLt GEMMs, 3 SiLU kernels and the coord_diff/radial
// elementwise ops (reference models/FastEGNN.py:144-150, 166-173, 237-246)
// — all [M,.] intermediates except the two outputs stay in LDS/registers.
//
// Mapping: 256 threads = 4 waves per block; each block owns a tile of 64
// row-sorted edges; each wave computes a 16-edge x 64-feature MFMA tile
// (mfma_f32_16x16x32_bf16, fp32 accumulation). Inputs are gathered into an
// LDS tile [64][K_STRIDE] (bank-conflict-padded); W1/W2/W3 are staged to
// LDS once per block in torch Linear layout [out][in] (which is exactly
// the B-fragment's k-contiguous layout for D = A @ W^T).

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int H = 64;          // hidden_nf (kernel is specialized for 64)
constexpr int EA = 2;          // edge_attr_nf
constexpr int K_IN = 2 * H + 1 + EA;  // 131
constexpr int K_PAD = 160;     // 5 MFMA k-steps of 32
constexpr int K_STRIDE = 168;  // LDS row stride (bank-conflict-free b128)
constexpr int H_STRIDE = 72;   // LDS stride for H-wide tiles
constexpr int TILE = 64;       // edges per block
constexpr int THREADS = 256;   // 4 waves

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ float silu(float x) {
  return x / (1.f + __expf(-x));
}
__device__ __forceinline__ float dsilu(float x) {  // d/dx SiLU(x)
  float s = 1.f / (1.f + __expf(-x));
  return s * (1.f + x * (1.f - s));
}

// ---- LDS layout (single dynamic allocation, 16B-aligned carves) ----------
struct SmemLayout {
  // offsets in bytes
  int in_tile;   // [TILE][K_STRIDE] bf16
  int w1;        // [H][K_STRIDE]   bf16 (torch [out][in], padded)
  int w2;        // [H][H_STRIDE]   bf16
  int w3;        // [H][H_STRIDE]   bf16
  int t1;        // [TILE][H_STRIDE] bf16
  int msg;       // [TILE][H_STRIDE] bf16
  int diff;      // [TILE][4] float (xyz + radial_raw)
  int pvec;      // [TILE] float
  int bias;      // [3*H + H] float (b1,b2,b3,w3v)
  int total;
};

__host__ __device__ constexpr SmemLayout smem_layout() {
  SmemLayout L{};
  int o = 0;
  L.in_tile = o; o += TILE * K_STRIDE * 2;
  L.w1 = o; o += H * K_STRIDE * 2;
  L.w2 = o; o += H * H_STRIDE * 2;
  L.w3 = o; o += H * H_STRIDE * 2;
  L.t1 = o; o += TILE * H_STRIDE * 2;
  L.msg = o; o += TILE * H_STRIDE * 2;
  L.diff = o; o += TILE * 4 * 4;
  L.pvec = o; o += TILE * 4;
  L.bias = o; o += 4 * H * 4;
  L.total = o;
  return L;
}

__device__ __forceinline__ bf16x8 lds_read8(const char* smem, int byte_off) {
  return *reinterpret_cast<const bf16x8*>(smem + byte_off);
}

// One 16(edge)x64(out) GEMM over LDS tiles: A [TILE][a_stride] bf16 rows
// a_row0.., B (weights) [64][b_stride] with k contiguous, ksteps of 32.
// acc[nt] — 4 accumulators of f32x4 (C layout col=l&15(+16nt),
// row=(l>>4)*4+r).
__device__ __forceinline__ void mfma_16x64(const char* smem, int a_off,
                                           int a_stride, int b_off,
                                           int b_stride, int ksteps,
                                           int lane, f32x4 acc[4]) {
  int arow = lane & 15;
  int kbase = (lane >> 4) * 8;
  for (int kk = 0; kk < ksteps; ++kk) {
    int k = kk * 32 + kbase;
    bf16x8 a = lds_read8(smem, a_off + arow * a_stride + k * 2);
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      bf16x8 b = lds_read8(smem,
                           b_off + (nt * 16 + (lane & 15)) * b_stride + k * 2);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
  }
}

// Stage a [H][in_w] weight matrix into LDS rows of byte-stride `stride`
// (zero-padding columns in_w..pad_w).
__device__ __forceinline__ void stage_weight(const bf16* __restrict__ w,
                                             char* smem, int off, int in_w,
                                             int pad_w, int stride, int tid) {
  for (int idx = tid; idx < H * pad_w / 8; idx += THREADS) {
    int r = idx / (pad_w / 8);
    int c8 = (idx % (pad_w / 8)) * 8;
    bf16x8 v = {};
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      int c = c8 + u;
      v[u] = (c < in_w) ? ((const __bf16*)w)[r * in_w + c] : (__bf16)0.f;
    }
    *reinterpret_cast<bf16x8*>(smem + off + r * stride + c8 * 2) = v;
  }
}

__global__ __launch_bounds__(THREADS) void fused_edge_fwd(
    const bf16* __restrict__ h,        // [N,64]
    const float* __restrict__ coord,   // [N,3]
    const float* __restrict__ eattr,   // [M,EA]
    const long* __restrict__ row,      // [M]
    const long* __restrict__ col,      // [M]
    const bf16* __restrict__ w1, const float* __restrict__ b1,
    const bf16* __restrict__ w2, const float* __restrict__ b2,
    const bf16* __restrict__ w3, const float* __restrict__ b3,
    const float* __restrict__ w3v,     // [64] head vector
    bf16* __restrict__ msg_out,        // [M,64]
    float* __restrict__ trans_out,     // [M,3]
    long m, int normalize, float eps) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr SmemLayout L = smem_layout();
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  // stage weights + biases once per block
  stage_weight(w1, smem, L.w1, K_IN, K_PAD, K_STRIDE * 2, tid);
  stage_weight(w2, smem, L.w2, H, H, H_STRIDE * 2, tid);
  stage_weight(w3, smem, L.w3, H, H, H_STRIDE * 2, tid);
  float* biases = reinterpret_cast<float*>(smem + L.bias);
  for (int i = tid; i < H; i += THREADS) {
    biases[i] = b1[i];
    biases[H + i] = b2[i];
    biases[2 * H + i] = b3[i];
    biases[3 * H + i] = w3v[i];
  }

  for (long tile = blockIdx.x; tile * TILE < m; tile += gridDim.x) {
    long e0 = tile * TILE;
    int nedge = (int)((m - e0 < (long)TILE) ? (m - e0) : (long)TILE);
    __syncthreads();  // protect LDS reuse across tiles

    // ---- gather stage: build in_tile [64][K_STRIDE] ----
    // 4 threads per edge: t covers h_i/h_j in 16B pieces.
    for (int idx = tid; idx < TILE * 16; idx += THREADS) {
      int e = idx / 16;       // edge in tile
      int piece = idx % 16;   // 16 x 16B pieces = 2 rows of 64 bf16
      char* dst = smem + L.in_tile + e * K_STRIDE * 2;
      if (e < nedge) {
        long ge = e0 + e;
        long src_node = piece < 8 ? row[ge] : col[ge];
        int c8 = (piece & 7) * 8;
        bf16x8 v = *reinterpret_cast<const bf16x8*>(h + src_node * H + c8);
        *reinterpret_cast<bf16x8*>(dst + (piece < 8 ? c8 : H + c8) * 2) = v;
      } else {
        int c8 = (piece & 7) * 8;
        bf16x8 z = {};
        *reinterpret_cast<bf16x8*>(dst + (piece < 8 ? c8 : H + c8) * 2) = z;
      }
    }
    // radial / coord_diff / edge_attr + zero K padding (1 thread per edge)
    for (int e = tid; e < TILE; e += THREADS) {
      char* dst = smem + L.in_tile + e * K_STRIDE * 2;
      float* dptr = reinterpret_cast<float*>(smem + L.diff) + e * 4;
      float dx = 0.f, dy = 0.f, dz = 0.f, r2 = 0.f, a0 = 0.f, a1 = 0.f;
      if (e < nedge) {
        long ge = e0 + e;
        long i = row[ge], j = col[ge];
        dx = coord[i * 3] - coord[j * 3];
        dy = coord[i * 3 + 1] - coord[j * 3 + 1];
        dz = coord[i * 3 + 2] - coord[j * 3 + 2];
        r2 = dx * dx + dy * dy + dz * dz;
        a0 = eattr[ge * EA];
        a1 = eattr[ge * EA + 1];
        if (normalize) {
          float inv = 1.f / (sqrtf(r2) + eps);
          dx *= inv; dy *= inv; dz *= inv;
        }
      }
      dptr[0] = dx; dptr[1] = dy; dptr[2] = dz; dptr[3] = r2;
      __bf16* brow = reinterpret_cast<__bf16*>(dst);
      brow[2 * H] = (__bf16)r2;
      brow[2 * H + 1] = (__bf16)a0;
      brow[2 * H + 2] = (__bf16)a1;
#pragma unroll
      for (int k = K_IN; k < K_PAD; ++k) brow[k] = (__bf16)0.f;
    }
    __syncthreads();

    // ---- GEMM1: t1 = SiLU(in @ W1^T + b1) ----
    {
      f32x4 acc[4] = {};
      mfma_16x64(smem, L.in_tile + wave * 16 * K_STRIDE * 2, K_STRIDE * 2,
                 L.w1, K_STRIDE * 2, K_PAD / 32, lane, acc);
      __bf16* t1 = reinterpret_cast<__bf16*>(smem + L.t1);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        int c = nt * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int e = wave * 16 + (lane >> 4) * 4 + r;
          t1[e * H_STRIDE + c] = (__bf16)silu(acc[nt][r] + biases[c]);
        }
      }
    }
    __syncthreads();

    // ---- GEMM2: msg = SiLU(t1 @ W2^T + b2) ----
    {
      f32x4 acc[4] = {};
      mfma_16x64(smem, L.t1 + wave * 16 * H_STRIDE * 2, H_STRIDE * 2, L.w2,
                 H_STRIDE * 2, H / 32, lane, acc);
      __bf16* mg = reinterpret_cast<__bf16*>(smem + L.msg);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        int c = nt * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int e = wave * 16 + (lane >> 4) * 4 + r;
          mg[e * H_STRIDE + c] = (__bf16)silu(acc[nt][r] + biases[H + c]);
        }
      }
    }
    __syncthreads();

    // ---- GEMM3 + head: p = SiLU(msg @ W3^T + b3) . w3v ----
    {
      f32x4 acc[4] = {};
      mfma_16x64(smem, L.msg + wave * 16 * H_STRIDE * 2, H_STRIDE * 2, L.w3,
                 H_STRIDE * 2, H / 32, lane, acc);
      float part[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) part[r] = 0.f;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        int c = nt * 16 + (lane & 15);
        float wv = biases[3 * H + c];
        float bb = biases[2 * H + c];
#pragma unroll
        for (int r = 0; r < 4; ++r)
          part[r] += silu(acc[nt][r] + bb) * wv;
      }
      // reduce the 16 lanes of each quarter-wave (same rows)
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) {
#pragma unroll
        for (int r = 0; r < 4; ++r)
          part[r] += __shfl_xor(part[r], off, 64);
      }
      if ((lane & 15) == 0) {
        float* pv = reinterpret_cast<float*>(smem + L.pvec);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int e = wave * 16 + (lane >> 4) * 4 + r;
          pv[e] = part[r];
        }
      }
    }
    __syncthreads();

    // ---- write msg (coalesced from LDS) + trans ----
    for (int idx = tid; idx < TILE * 8; idx += THREADS) {
      int e = idx / 8;
      if (e >= nedge) continue;
      int c8 = (idx % 8) * 8;
      bf16x8 v = lds_read8(smem, L.msg + (e * H_STRIDE + c8) * 2);
      *reinterpret_cast<bf16x8*>(msg_out + (e0 + e) * H + c8) = v;
    }
    for (int e = tid; e < nedge; e += THREADS) {
      const float* dptr = reinterpret_cast<float*>(smem + L.diff) + e * 4;
      float p = reinterpret_cast<float*>(smem + L.pvec)[e];
      trans_out[(e0 + e) * 3] = dptr[0] * p;
      trans_out[(e0 + e) * 3 + 1] = dptr[1] * p;
      trans_out[(e0 + e) * 3 + 2] = dptr[2] * p;
    }
  }
}

// ---- MFMA layout probe (test harness): D = A[16x32] @ B[32x16] ----------
__global__ void mfma_probe_kernel(const bf16* __restrict__ a,   // [16][32]
                                  const bf16* __restrict__ bt,  // [16][32] B^T
                                  float* __restrict__ d) {      // [16][16]
  int lane = threadIdx.x & 63;
  bf16x8 av, bv;
#pragma unroll
  for (int u = 0; u < 8; ++u) {
    int r = lane & 15, k = (lane >> 4) * 8 + u;
    av[u] = ((const __bf16*)a)[r * 32 + k];
    bv[u] = ((const __bf16*)bt)[r * 32 + k];  // bt[col][k] = B[k][col]
  }
  f32x4 acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bv, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int rr = (lane >> 4) * 4 + r, cc = lane & 15;
    d[rr * 16 + cc] = acc[r];
  }
}

}  // namespace

std::tuple<torch::Tensor, torch::Tensor> fused_edge_forward(
    torch::Tensor h, torch::Tensor coord, torch::Tensor eattr,
    torch::Tensor row, torch::Tensor col, torch::Tensor w1, torch::Tensor b1,
    torch::Tensor w2, torch::Tensor b2, torch::Tensor w3, torch::Tensor b3,
    torch::Tensor w3v, bool normalize, double eps) {
  TORCH_CHECK(h.is_cuda() && h.scalar_type() == torch::kBFloat16,
              "h must be CUDA bf16");
  TORCH_CHECK(h.size(1) == H, "fused edge kernel requires hidden_nf=64");
  TORCH_CHECK(eattr.size(1) == EA, "fused edge kernel requires edge_attr_nf=2");
  auto hc = h.contiguous();
  auto cc = coord.contiguous().to(torch::kFloat);
  auto ec = eattr.contiguous().to(torch::kFloat);
  long m = row.numel();
  auto msg = torch::empty({m, (long)H}, h.options());
  auto trans = torch::empty({m, 3}, cc.options());
  if (m == 0) return {msg, trans};
  auto stream = at::hip::getCurrentHIPStream();
  constexpr SmemLayout L = smem_layout();
  long tiles = (m + TILE - 1) / TILE;
  int blocks = (int)std::min<long>(tiles, 8192);
  auto w1c = w1.contiguous();
  auto w2c = w2.contiguous();
  auto w3c = w3.contiguous();
  auto b1c = b1.contiguous().to(torch::kFloat);
  auto b2c = b2.contiguous().to(torch::kFloat);
  auto b3c = b3.contiguous().to(torch::kFloat);
  auto w3vc = w3v.contiguous().to(torch::kFloat);
  fused_edge_fwd<<<blocks, THREADS, L.total, stream>>>(
      reinterpret_cast<const bf16*>(hc.data_ptr()), cc.data_ptr<float>(),
      ec.data_ptr<float>(), row.contiguous().data_ptr<long>(),
      col.contiguous().data_ptr<long>(),
      reinterpret_cast<const bf16*>(w1c.data_ptr()), b1c.data_ptr<float>(),
      reinterpret_cast<const bf16*>(w2c.data_ptr()), b2c.data_ptr<float>(),
      reinterpret_cast<const bf16*>(w3c.data_ptr()), b3c.data_ptr<float>(),
      w3vc.data_ptr<float>(),
      reinterpret_cast<bf16*>(msg.data_ptr()), trans.data_ptr<float>(), m,
      normalize ? 1 : 0, (float)eps);
  return {msg, trans};
}

torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor bt) {
  auto ac = a.contiguous();
  auto btc = bt.contiguous();
  auto d = torch::empty({16, 16},
                        a.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  mfma_probe_kernel<<<1, 64, 0, stream>>>(
      reinterpret_cast<const bf16*>(ac.data_ptr()),
      reinterpret_cast<const bf16*>(btc.data_ptr()), d.data_ptr<float>());
  return d;
}
