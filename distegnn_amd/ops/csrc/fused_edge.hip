// Fused FastEGNN edge block — the north-star MFMA kernel (gfx950).
//
// Per edge e = (i=row[e], j=col[e]) this computes, in ONE kernel:
//   in_e   = [h_i | h_j | r_e | a_e]            (K = 2H+1+Ea)
//   r_e    = ||x_i - x_j||^2, d_e = x_i - x_j   (optionally normalized)
//   t1     = SiLU(in_e  @ W1^T + b1)            [H]
//   msg    = SiLU(t1    @ W2^T + b2)            [H]   (edge_feat, phi_e)
//   s3     = SiLU(msg   @ W3^T + b3)            [H]   (phi_x hidden)
//   p_e    = s3 . w3                            scalar (phi_x head)
//   trans  = d_e * p_e                          [3]
// writing msg [M,H] bf16 + trans [M,3] f32 (aggregated by the CSR
// segment-mean kernels). All [M,.] intermediates stay in LDS/registers.
//
// Templated over H in {32, 64, 128} (round 2): tile sizes, LDS strides
// and MFMA n-tile counts derive from H; H=64 is the tuned headline shape
// (the numbers in profiles/ are measured there), H=32/128 make any
// hidden_nf the YAML surface allows run the MFMA path instead of
// silently dropping to eager.
//
// Occupancy design (PMC-driven): weights (<=40 KB at H=64, L2-resident,
// shared by every block) are read DIRECTLY from global memory from
// row-padded copies prepared in the launcher; LDS holds only the
// per-tile input/activation buffers. Wave-per-subtile execution: every
// phase reads/writes only its own wave's 16 rows, so the tile loop
// carries NO barriers.

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int EA = 2;
constexpr int TILE = 64;
constexpr int THREADS = 256;

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ float silu(float x) {
  return x / (1.f + __expf(-x));
}

// per-H derived dimensions (K_PAD multiple of the 32-wide MFMA k-step;
// +8-element strides keep the 16 b128 fragment readers on distinct banks)
template <int H>
struct ED {
  static constexpr int K_IN = 2 * H + 1 + EA;
  static constexpr int K_PAD = (K_IN + 31) / 32 * 32;
  static constexpr int K_STRIDE = K_PAD + 8;
  static constexpr int H_STRIDE = H + 8;
  static constexpr int NT = H / 16;     // 16-col MFMA n-tiles per row
  static constexpr int HP = H / 8;      // bf16x8 fragments per H row
};

template <int H>
struct SmemLayout {
  int in_tile;  // [TILE][K_STRIDE] bf16
  int t1;       // [TILE][H_STRIDE] bf16
  int msg;      // [TILE][H_STRIDE] bf16
  int diff;     // [TILE][4] f32
  int pvec;     // [TILE] f32
  int bias;     // [4*H] f32
  int rows;     // [TILE] i32 (edge endpoints, staged once per tile)
  int cols;     // [TILE] i32
  int total;
};

template <int H>
__host__ __device__ constexpr SmemLayout<H> smem_layout() {
  SmemLayout<H> L{};
  int o = 0;
  L.in_tile = o; o += TILE * ED<H>::K_STRIDE * 2;
  L.t1 = o; o += TILE * ED<H>::H_STRIDE * 2;
  L.msg = o; o += TILE * ED<H>::H_STRIDE * 2;
  L.diff = o; o += TILE * 4 * 4;
  L.pvec = o; o += TILE * 4;
  L.bias = o; o += 4 * H * 4;
  L.rows = o; o += TILE * 4;
  L.cols = o; o += TILE * 4;
  L.total = o;
  return L;
}

__device__ __forceinline__ bf16x8 lds8(const char* smem, int off) {
  return *reinterpret_cast<const bf16x8*>(smem + off);
}
__device__ __forceinline__ bf16x8 g8(const bf16* p) {
  return *reinterpret_cast<const bf16x8*>(p);
}
// prevent LICM from hoisting all phases' weight fragments into registers
__device__ __forceinline__ const bf16* opaque(const bf16* p) {
  asm volatile("" : "+v"(p));
  return p;
}

// 16(edges) x H GEMM: A from LDS, B (weights, k-contig rows [H][kb])
// from GLOBAL (L2-resident).
template <int KSTEPS, int NT>
__device__ __forceinline__ void mm_a_lds(const char* smem, int a_off,
                                         int a_stride,
                                         const bf16* __restrict__ w, int wk,
                                         int lane, f32x4 (&acc)[NT]) {
__builtin_amdgcn_s_setprio(1);
  #pragma unroll
  for (int kk = 0; kk < KSTEPS; ++kk) {
    int k = kk * 32 + (lane >> 4) * 8;
    bf16x8 a = lds8(smem, a_off + (lane & 15) * a_stride + k * 2);
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) {
      bf16x8 b = g8(w + (nt * 16 + (lane & 15)) * wk + k);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
  }
  __builtin_amdgcn_s_setprio(0);
}

template <int H>
__global__ __launch_bounds__(THREADS, H <= 64 ? 2 : 1) void fused_edge_fwd(
    const bf16* __restrict__ h, const float* __restrict__ coord,
    const float* __restrict__ eattr, const long* __restrict__ row,
    const long* __restrict__ col,
    const bf16* __restrict__ w1p,  // [H][K_PAD] padded
    const float* __restrict__ b1, const bf16* __restrict__ w2,
    const float* __restrict__ b2, const bf16* __restrict__ w3,
    const float* __restrict__ b3, const float* __restrict__ w3v,
    bf16* __restrict__ msg_out, float* __restrict__ trans_out, long m,
    int normalize, float eps) {
  constexpr int K_IN = ED<H>::K_IN;
  constexpr int K_PAD = ED<H>::K_PAD;
  constexpr int K_STRIDE = ED<H>::K_STRIDE;
  constexpr int H_STRIDE = ED<H>::H_STRIDE;
  constexpr int NT = ED<H>::NT;
  constexpr int HP = ED<H>::HP;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr SmemLayout<H> L = smem_layout<H>();
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  float* biases = reinterpret_cast<float*>(smem + L.bias);
  for (int i = tid; i < H; i += THREADS) {
    biases[i] = b1[i];
    biases[H + i] = b2[i];
    biases[2 * H + i] = b3[i];
    biases[3 * H + i] = w3v[i];
  }
  __syncthreads();  // biases initialized by wave 0

  // Wave-per-subtile execution: every phase reads/writes only its own
  // wave's 16 rows, so the tile loop carries NO barriers.
  //
  // XCD-aware tile remap: the dispatcher places workgroup b on XCD b%8
  // (each XCD has a private 4 MiB L2). In identity order neighboring
  // 64-edge tiles — whose src-sorted row gathers share h/x rows
  // (~14.6 edges/node) — land on DIFFERENT XCDs, so every XCD streams
  // the whole 15 MB feature table. The bijective remap below gives each
  // XCD a CONTIGUOUS tile range (~1/8 of the rows ≈ 1.9 MB, L2-fits).
  const long ntile = (m + TILE - 1) / TILE;
  const long tq = ntile >> 3, tr = ntile & 7;
  for (long vt = blockIdx.x; vt < ntile; vt += gridDim.x) {
    const long xcd = vt & 7, ti = vt >> 3;
    const long tile =
        (xcd < tr ? xcd * (tq + 1) : tr * (tq + 1) + (xcd - tr) * tq) + ti;
    long e0 = tile * TILE;
    int nedge = (int)((m - e0 < (long)TILE) ? (m - e0) : (long)TILE);
    int* rws = reinterpret_cast<int*>(smem + L.rows);
    int* cls = reinterpret_cast<int*>(smem + L.cols);
    if (lane < 16) {
      int e = wave * 16 + lane;
      rws[e] = e < nedge ? (int)row[e0 + e] : 0;
      cls[e] = e < nedge ? (int)col[e0 + e] : 0;
    }

    // gather stage: in_tile [64][K_STRIDE] (wave-local rows)
    for (int idx = lane; idx < 16 * 2 * HP; idx += 64) {
      int e = wave * 16 + idx / (2 * HP), piece = idx % (2 * HP);
      char* dst = smem + L.in_tile + e * K_STRIDE * 2;
      int c8 = (piece % HP) * 8;
      bf16x8 v = {};
      if (e < nedge) {
        long src = piece < HP ? rws[e] : cls[e];
        v = g8(h + src * H + c8);
      }
      *reinterpret_cast<bf16x8*>(dst + (piece < HP ? c8 : H + c8) * 2) = v;
    }
    if (lane < 16) {
      int e = wave * 16 + lane;
      char* dst = smem + L.in_tile + e * K_STRIDE * 2;
      float* dptr = reinterpret_cast<float*>(smem + L.diff) + e * 4;
      float dx = 0, dy = 0, dz = 0, r2 = 0, a0 = 0, a1 = 0;
      if (e < nedge) {
        long ge = e0 + e;
        long i = rws[e], j = cls[e];
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

    {  // GEMM1 -> t1
      f32x4 acc[NT] = {};
      mm_a_lds<K_PAD / 32, NT>(smem, L.in_tile + wave * 16 * K_STRIDE * 2,
                               K_STRIDE * 2, opaque(w1p), K_PAD, lane, acc);
      __bf16* t1 = reinterpret_cast<__bf16*>(smem + L.t1);
#pragma unroll
      for (int nt = 0; nt < NT; ++nt) {
        int c = nt * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int e = wave * 16 + (lane >> 4) * 4 + r;
          t1[e * H_STRIDE + c] = (__bf16)silu(acc[nt][r] + biases[c]);
        }
      }
    }
    {  // GEMM2 -> msg
      f32x4 acc[NT] = {};
      mm_a_lds<H / 32, NT>(smem, L.t1 + wave * 16 * H_STRIDE * 2,
                           H_STRIDE * 2, opaque(w2), H, lane, acc);
      __bf16* mg = reinterpret_cast<__bf16*>(smem + L.msg);
#pragma unroll
      for (int nt = 0; nt < NT; ++nt) {
        int c = nt * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int e = wave * 16 + (lane >> 4) * 4 + r;
          mg[e * H_STRIDE + c] = (__bf16)silu(acc[nt][r] + biases[H + c]);
        }
      }
    }
    {  // GEMM3 + head -> p
      f32x4 acc[NT] = {};
      mm_a_lds<H / 32, NT>(smem, L.msg + wave * 16 * H_STRIDE * 2,
                           H_STRIDE * 2, opaque(w3), H, lane, acc);
      float part[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int nt = 0; nt < NT; ++nt) {
        int c = nt * 16 + (lane & 15);
        float wv = biases[3 * H + c];
        float bb = biases[2 * H + c];
#pragma unroll
        for (int r = 0; r < 4; ++r) part[r] += silu(acc[nt][r] + bb) * wv;
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
#pragma unroll
        for (int r = 0; r < 4; ++r) part[r] += __shfl_xor(part[r], off, 64);
      if ((lane & 15) == 0) {
        float* pv = reinterpret_cast<float*>(smem + L.pvec);
#pragma unroll
        for (int r = 0; r < 4; ++r)
          pv[wave * 16 + (lane >> 4) * 4 + r] = part[r];
      }
    }

    for (int idx = lane; idx < 16 * HP; idx += 64) {
      int e = wave * 16 + idx / HP;
      if (e >= nedge) continue;
      int c8 = (idx % HP) * 8;
      *reinterpret_cast<bf16x8*>(msg_out + (e0 + e) * H + c8) =
          lds8(smem, L.msg + (e * H_STRIDE + c8) * 2);
    }
    if (lane < 16 && wave * 16 + lane < nedge) {
      int e = wave * 16 + lane;
      const float* dptr = reinterpret_cast<float*>(smem + L.diff) + e * 4;
      float p = reinterpret_cast<float*>(smem + L.pvec)[e];
      trans_out[(e0 + e) * 3] = dptr[0] * p;
      trans_out[(e0 + e) * 3 + 1] = dptr[1] * p;
      trans_out[(e0 + e) * 3 + 2] = dptr[2] * p;
    }
  }
}

// ---- MFMA layout probe (test harness): D = A[16x32] @ B[32x16] ----------
__global__ void mfma_probe_kernel(const bf16* __restrict__ a,
                                  const bf16* __restrict__ bt,
                                  float* __restrict__ d) {
  int lane = threadIdx.x & 63;
  bf16x8 av, bv;
#pragma unroll
  for (int u = 0; u < 8; ++u) {
    int r = lane & 15, k = (lane >> 4) * 8 + u;
    av[u] = ((const __bf16*)a)[r * 32 + k];
    bv[u] = ((const __bf16*)bt)[r * 32 + k];
  }
  f32x4 acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bv, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r)
    d[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

}  // namespace

std::tuple<torch::Tensor, torch::Tensor> fused_edge_forward(
    torch::Tensor h, torch::Tensor coord, torch::Tensor eattr,
    torch::Tensor row, torch::Tensor col, torch::Tensor w1, torch::Tensor b1,
    torch::Tensor w2, torch::Tensor b2, torch::Tensor w3, torch::Tensor b3,
    torch::Tensor w3v, bool normalize, double eps,
    std::vector<torch::Tensor> prepped) {
  TORCH_CHECK(h.is_cuda() && h.scalar_type() == torch::kBFloat16,
              "h must be CUDA bf16");
  long hdim = h.size(1);
  TORCH_CHECK(hdim == 32 || hdim == 64 || hdim == 128,
              "fused edge kernel supports hidden_nf in {32, 64, 128}");
  TORCH_CHECK(eattr.size(1) == EA, "fused edge kernel requires edge_attr_nf=2");
  auto hc = h.contiguous();
  auto cc = coord.contiguous().to(torch::kFloat);
  auto ec = eattr.contiguous().to(torch::kFloat);
  long m = row.numel();
  auto msg = torch::empty({m, hdim}, h.options());
  auto trans = torch::empty({m, 3}, cc.options());
  if (m == 0) return {msg, trans};
  auto stream = at::hip::getCurrentHIPStream();
  long tiles = (m + TILE - 1) / TILE;
  int blocks = (int)std::min<long>(tiles, 16384);
  // Weight prep: python may pass version-cached transformed copies
  // (ops/prep.py) so replays/graphs skip ~7 tiny cast/pad kernels per
  // call; otherwise transform here.
  torch::Tensor w1p, w2c, w3c, b1c, b2c, b3c, w3vc;
  long k_in = 2 * hdim + 1 + EA;
  long k_pad = (k_in + 31) / 32 * 32;
  if (!prepped.empty()) {
    TORCH_CHECK(prepped.size() == 7, "edge fwd prepped wants 7 tensors");
    w1p = prepped[0]; w2c = prepped[1]; w3c = prepped[2];
    b1c = prepped[3]; b2c = prepped[4]; b3c = prepped[5]; w3vc = prepped[6];
    TORCH_CHECK(w1p.size(1) == k_pad, "prepped w1p must be row-padded");
  } else {
    // row-pad W1 to [H][K_PAD] so 16-B B-fragment reads are aligned
    w1p = torch::constant_pad_nd(w1.contiguous(), {0, k_pad - k_in});
    w2c = w2.contiguous();
    w3c = w3.contiguous();
    b1c = b1.contiguous().to(torch::kFloat);
    b2c = b2.contiguous().to(torch::kFloat);
    b3c = b3.contiguous().to(torch::kFloat);
    w3vc = w3v.contiguous().to(torch::kFloat);
  }
#define LAUNCH_FWD(HH)                                                      \
  fused_edge_fwd<HH><<<blocks, THREADS, smem_layout<HH>().total, stream>>>( \
      reinterpret_cast<const bf16*>(hc.data_ptr()), cc.data_ptr<float>(),   \
      ec.data_ptr<float>(), row.contiguous().data_ptr<long>(),              \
      col.contiguous().data_ptr<long>(),                                    \
      reinterpret_cast<const bf16*>(w1p.data_ptr()), b1c.data_ptr<float>(), \
      reinterpret_cast<const bf16*>(w2c.data_ptr()), b2c.data_ptr<float>(), \
      reinterpret_cast<const bf16*>(w3c.data_ptr()), b3c.data_ptr<float>(), \
      w3vc.data_ptr<float>(),                                               \
      reinterpret_cast<bf16*>(msg.data_ptr()), trans.data_ptr<float>(), m,  \
      normalize ? 1 : 0, (float)eps)
  if (hdim == 32) LAUNCH_FWD(32);
  else if (hdim == 64) LAUNCH_FWD(64);
  else LAUNCH_FWD(128);
#undef LAUNCH_FWD
  return {msg, trans};
}

torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor bt) {
  auto ac = a.contiguous();
  auto btc = bt.contiguous();
  auto d = torch::empty({16, 16}, a.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  mfma_probe_kernel<<<1, 64, 0, stream>>>(
      reinterpret_cast<const bf16*>(ac.data_ptr()),
      reinterpret_cast<const bf16*>(btc.data_ptr()), d.data_ptr<float>());
  return d;
}
