// Fused FastEGNN virtual-edge block (gfx950).
//
// Rows are (node n, virtual channel c) pairs, R = N*C. Per row:
//   vdiff  = X[b,c] - x_n             (b = batch[n]; [3], fp32)
//   vrad   = ||vdiff||
//   vin    = [h_n | Z[b,c] | vrad | gram[b,c,:]]   (K = 129+C <= 137)
//   t1     = SiLU(vin @ W1^T + b1);  vmsg = SiLU(t1 @ W2^T + b2)  [64]
//   pxv    = silu(vmsg @ Wxv^T + bxv) . wxv     (phi_xv head)
//   pX     = silu(vmsg @ WX^T  + bX ) . wX      (phi_X head)
//   tv     = -vdiff * pxv ;  tx = vdiff * pX    ([3] each)
// The model consumes vmsg (mean over c + per-graph pools), tv (mean over
// c -> coordinate update) and tx (per-graph pool -> virtual coordinates).
// This replaces the eager chain's [N,C,3] vdiff/vradial materialization,
// the [R,129+C] concat, four Linear forwards and the head elementwise ops
// (reference models/FastEGNN.py:154-163, 180, 192, 252-253, 266).
//
// TRAIN=1 additionally writes the per-row activations/pre-activations the
// backward needs (vin, z1, z2, zxv, zX, p2). The backward kernel consumes
// those (no recompute) and emits dz tensors for python-side split-K wgrad
// GEMMs plus per-row input grads (dh, dvfeat, dgram, dvd) that reduce via
// sum-over-c / per-graph pools. Occupancy design follows fused_edge.hip:
// weights read from L2-resident padded/transposed global copies, LDS holds
// only per-tile buffers, pointer-laundering bounds fragment liveness.

#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include "common.h"

namespace {

constexpr int H = 64;
constexpr int K_PAD = 160;
constexpr int K_STRIDE = 168;
constexpr int K_OUT = 144;
constexpr int H_STRIDE = 72;
constexpr int TILE = 64;
constexpr int THREADS = 256;
constexpr int CMAX = 8;

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ float silu_(float x) {
  return x / (1.f + __expf(-x));
}
__device__ __forceinline__ float dsilu_(float x) {
  float s = 1.f / (1.f + __expf(-x));
  return s * (1.f + x * (1.f - s));
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
__device__ __forceinline__ bf16x8 lds8_silu(const char* smem, int off) {
  bf16x8 z = lds8(smem, off);
  bf16x8 r;
#pragma unroll
  for (int u = 0; u < 8; ++u) r[u] = (__bf16)silu_((float)z[u]);
  return r;
}

template <int KSTEPS, bool SILU_A>
__device__ __forceinline__ void mm_g(const char* smem, int a_off,
                                     int a_stride, const bf16* w, int wk,
                                     int lane, f32x4 (&acc)[4]) {
  __builtin_amdgcn_s_setprio(1);
#pragma unroll
  for (int kk = 0; kk < KSTEPS; ++kk) {
    int k = kk * 32 + (lane >> 4) * 8;
    bf16x8 a = SILU_A ? lds8_silu(smem, a_off + (lane & 15) * a_stride + k * 2)
                      : lds8(smem, a_off + (lane & 15) * a_stride + k * 2);
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      bf16x8 b = g8(w + (nt * 16 + (lane & 15)) * wk + k);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
    }
  }
  __builtin_amdgcn_s_setprio(0);
}

struct VSmem {
  int in_tile;  // [TILE][K_STRIDE] bf16 (bwd also time-shares it for the
                //   zX/z2 stagings and the dzX tile -> 3 blocks/CU)
  int za;       // [TILE][H_STRIDE] bf16 (z1; bwd: stagings/dz1)
  int zb;       // [TILE][H_STRIDE] bf16 (z2; bwd: dz2)
  int zc;       // [TILE][H_STRIDE] bf16 (zxv / zX time-shared; bwd: dzxv)
  int diff;     // [TILE][4] f32 (vdiff xyz, vrad)
  int scal;     // [TILE][4] f32 (pxv, pX / dpxv, dr)
  int bias;     // [6*H] f32 (b1, b2, bxv, bX, wxv, wX)
  int gbacc;    // [6*H] f32 (bwd: gb1, gb2, gbxv, gbX, gwxvv, gwXv)
  int zd;       // end marker: kernels allocate smem up to here (zd itself
                //   is no longer a live region)
  int total;
};

__host__ __device__ constexpr VSmem vsmem_layout() {
  VSmem L{};
  int o = 0;
  L.in_tile = o; o += TILE * K_STRIDE * 2;
  L.za = o; o += TILE * H_STRIDE * 2;
  L.zb = o; o += TILE * H_STRIDE * 2;
  L.zc = o; o += TILE * H_STRIDE * 2;
  L.diff = o; o += TILE * 4 * 4;
  L.scal = o; o += TILE * 4 * 4;
  L.bias = o; o += 6 * H * 4;
  L.gbacc = o; o += 6 * H * 4;
  L.zd = o;
  L.total = o;
  return L;
}

// Backward-only layout: the staging region (sa) is H_STRIDE-wide — the
// dvin product streams through it in three 48-column passes instead of
// needing a K_STRIDE-wide tile — and the bias slot holds only wxvv/wXv.
// Total is exactly 40960 B/block -> 4 blocks/CU (the 122-VGPR backward
// allows 4 waves/SIMD; the full-width forward layout caps both at 3).
struct VSmemB {
  int sa;     // [TILE][H_STRIDE] bf16 (zX staging -> dzX -> z2 staging ->
              //   dvin pass window)
  int za;     // [TILE][H_STRIDE] bf16
  int zb;     // [TILE][H_STRIDE] bf16
  int zc;     // [TILE][H_STRIDE] bf16
  int diff;   // [TILE][4] f32
  int scal;   // [TILE][4] f32
  int bias;   // [2*H] f32 (wxvv | wXv)
  int gbacc;  // [6*H] f32
  int total;
};

__host__ __device__ constexpr VSmemB vsmem_layout_bwd() {
  VSmemB L{};
  int o = 0;
  L.sa = o; o += TILE * H_STRIDE * 2;
  L.za = o; o += TILE * H_STRIDE * 2;
  L.zb = o; o += TILE * H_STRIDE * 2;
  L.zc = o; o += TILE * H_STRIDE * 2;
  L.diff = o; o += TILE * 4 * 4;
  L.scal = o; o += TILE * 4 * 4;
  L.bias = o; o += 2 * H * 4;
  L.gbacc = o; o += 6 * H * 4;
  L.total = o;
  return L;
}

// C-layout epilogue write into an LDS bf16 tile
#define EPI_WRITE(tileptr, expr)                                            \
  _Pragma("unroll") for (int nt = 0; nt < 4; ++nt) {                        \
    int cc = nt * 16 + (lane & 15);                                         \
    _Pragma("unroll") for (int rr = 0; rr < 4; ++rr) {                      \
      int e = wave * 16 + (lane >> 4) * 4 + rr;                             \
      float x = acc[nt][rr];                                                \
      (tileptr)[e * H_STRIDE + cc] = (__bf16)(expr);                        \
    }                                                                       \
  }

// copy an LDS H-tile to global [R,64] bf16 (optionally through silu).
// Wave-local rows (wave-per-subtile execution, round 2): each wave copies
// only its own 16 rows, so no barrier is needed around the copy.
#define TILE_TO_GLOBAL(off, dst, SILU)                                      \
  for (int idx = lane; idx < 16 * 8; idx += 64) {                           \
    int e = wave * 16 + idx / 8;                                            \
    if (e >= nrow) continue;                                                \
    int c8 = (idx % 8) * 8;                                                 \
    *reinterpret_cast<bf16x8*>((dst) + (r0 + e) * H + c8) =                 \
        SILU ? lds8_silu(smem, (off) + (e * H_STRIDE + c8) * 2)             \
             : lds8(smem, (off) + (e * H_STRIDE + c8) * 2);                 \
  }

template <bool TRAIN>
__global__ __launch_bounds__(THREADS) void fused_virtual_fwd(
    const bf16* __restrict__ h,        // [N,64]
    const float* __restrict__ coord,   // [N,3]
    const float* __restrict__ vcoord,  // [B,C,3]
    const bf16* __restrict__ vfeat,    // [B,C,64]
    const float* __restrict__ gram,    // [B,C,C]
    const long* __restrict__ batch,    // [N]
    const bf16* __restrict__ w1p,      // [64][K_PAD]
    const bf16* __restrict__ w2, const bf16* __restrict__ wxv,
    const bf16* __restrict__ wX,
    const float* __restrict__ b1, const float* __restrict__ b2,
    const float* __restrict__ bxv, const float* __restrict__ bX,
    const float* __restrict__ wxvv, const float* __restrict__ wXv,
    bf16* __restrict__ vmsg_out,       // [R,64]
    float* __restrict__ tv_out,        // [R,3]
    float* __restrict__ tx_out,        // [R,3]
    bf16* __restrict__ vin_out,        // [R,K_OUT]   (TRAIN)
    bf16* __restrict__ z1_out, bf16* __restrict__ z2_out,
    bf16* __restrict__ zxv_out, bf16* __restrict__ zX_out,  // [R,64] (TRAIN)
    float* __restrict__ p2_out,        // [R,2]       (TRAIN)
    long n_rows, int cdim, int k_in) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr VSmem L = vsmem_layout();
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  float* biases = reinterpret_cast<float*>(smem + L.bias);
  for (int i = tid; i < H; i += THREADS) {
    biases[i] = b1[i];
    biases[H + i] = b2[i];
    biases[2 * H + i] = bxv[i];
    biases[3 * H + i] = bX[i];
    biases[4 * H + i] = wxvv[i];
    biases[5 * H + i] = wXv[i];
  }
  __syncthreads();  // biases initialized by wave 0

  // Wave-per-subtile execution (round 2): every phase touches only its
  // own wave's 16 rows -> no intra-tile barriers.
  for (long tile = blockIdx.x; tile * TILE < n_rows; tile += gridDim.x) {
    long r0 = tile * TILE;
    int nrow = (int)((n_rows - r0 < (long)TILE) ? (n_rows - r0)
                                                : (long)TILE);

    // ---- gather vin (wave-local rows) ----
    for (int idx = lane; idx < 16 * 16; idx += 64) {
      int e = wave * 16 + idx / 16, piece = idx % 16;
      char* dst = smem + L.in_tile + e * K_STRIDE * 2;
      int c8 = (piece & 7) * 8;
      bf16x8 v = {};
      if (e < nrow) {
        long r = r0 + e;
        long n = r / cdim, c = r - n * cdim;
        if (piece < 8) {
          v = g8(h + n * H + c8);
        } else {
          long b = batch[n];
          v = g8(vfeat + (b * cdim + c) * H + c8);
        }
      }
      *reinterpret_cast<bf16x8*>(dst + (piece < 8 ? c8 : H + c8) * 2) = v;
    }
    if (lane < 16) {
      int e = wave * 16 + lane;
      char* dst = smem + L.in_tile + e * K_STRIDE * 2;
      float* dptr = reinterpret_cast<float*>(smem + L.diff) + e * 4;
      __bf16* brow = reinterpret_cast<__bf16*>(dst);
      float dx = 0, dy = 0, dz = 0, vr = 0;
      if (e < nrow) {
        long r = r0 + e;
        long n = r / cdim, c = r - n * cdim;
        long b = batch[n];
        const float* X = vcoord + (b * cdim + c) * 3;
        dx = X[0] - coord[n * 3];
        dy = X[1] - coord[n * 3 + 1];
        dz = X[2] - coord[n * 3 + 2];
        vr = sqrtf(dx * dx + dy * dy + dz * dz);
        brow[2 * H] = (__bf16)vr;
        const float* gr = gram + (b * cdim + c) * cdim;
        for (int j = 0; j < cdim; ++j) brow[2 * H + 1 + j] = (__bf16)gr[j];
      } else {
        brow[2 * H] = (__bf16)0.f;
        for (int j = 0; j < cdim; ++j) brow[2 * H + 1 + j] = (__bf16)0.f;
      }
      dptr[0] = dx; dptr[1] = dy; dptr[2] = dz; dptr[3] = vr;
#pragma unroll
      for (int k = 0; k < K_PAD - 2 * H - 1 - CMAX; ++k)
        brow[2 * H + 1 + CMAX + k] = (__bf16)0.f;
      for (int j = cdim; j < CMAX; ++j) brow[2 * H + 1 + j] = (__bf16)0.f;
    }
    if (TRAIN) {
      for (int idx = lane; idx < 16 * (K_OUT / 8); idx += 64) {
        int e = wave * 16 + idx / (K_OUT / 8);
        if (e >= nrow) continue;
        int c8 = (idx % (K_OUT / 8)) * 8;
        *reinterpret_cast<bf16x8*>(vin_out + (r0 + e) * K_OUT + c8) =
            lds8(smem, L.in_tile + (e * K_STRIDE + c8) * 2);
      }
    }

    {  // z1
      f32x4 acc[4] = {};
      mm_g<K_PAD / 32, false>(smem, L.in_tile + wave * 16 * K_STRIDE * 2,
                              K_STRIDE * 2, opaque(w1p), K_PAD, lane, acc);
      __bf16* za = reinterpret_cast<__bf16*>(smem + L.za);
      EPI_WRITE(za, x + biases[cc]);
    }
    if (TRAIN) TILE_TO_GLOBAL(L.za, z1_out, false);
    {  // z2
      f32x4 acc[4] = {};
      mm_g<2, true>(smem, L.za + wave * 16 * H_STRIDE * 2, H_STRIDE * 2,
                    opaque(w2), H, lane, acc);
      __bf16* zb = reinterpret_cast<__bf16*>(smem + L.zb);
      EPI_WRITE(zb, x + biases[H + cc]);
    }
    if (TRAIN) TILE_TO_GLOBAL(L.zb, z2_out, false);
    TILE_TO_GLOBAL(L.zb, vmsg_out, true);

    // heads: zxv then zX, p reductions
#pragma unroll
    for (int head = 0; head < 2; ++head) {
      f32x4 acc[4] = {};
      mm_g<2, true>(smem, L.zb + wave * 16 * H_STRIDE * 2, H_STRIDE * 2,
                    opaque(head == 0 ? wxv : wX), H, lane, acc);
      __bf16* zt = reinterpret_cast<__bf16*>(smem + L.zc);
      float part[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        int cc = nt * 16 + (lane & 15);
        float bb = biases[(2 + head) * H + cc];
        float wv = biases[(4 + head) * H + cc];
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          int e = wave * 16 + (lane >> 4) * 4 + rr;
          float z = acc[nt][rr] + bb;
          zt[e * H_STRIDE + cc] = (__bf16)z;
          part[rr] += silu_(z) * wv;
        }
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
#pragma unroll
        for (int rr = 0; rr < 4; ++rr)
          part[rr] += __shfl_xor(part[rr], off, 64);
      if ((lane & 15) == 0) {
        float* sc = reinterpret_cast<float*>(smem + L.scal);
#pragma unroll
        for (int rr = 0; rr < 4; ++rr)
          sc[(wave * 16 + (lane >> 4) * 4 + rr) * 4 + head] = part[rr];
      }
      if (TRAIN) {
        if (head == 0) {
          TILE_TO_GLOBAL(L.zc, zxv_out, false);
        } else {
          TILE_TO_GLOBAL(L.zc, zX_out, false);
        }
      }
    }

    // tv / tx / p2 (wave-local rows)
    if (lane < 16 && wave * 16 + lane < nrow) {
      int e = wave * 16 + lane;
      const float* dptr = reinterpret_cast<float*>(smem + L.diff) + e * 4;
      const float* sc = reinterpret_cast<float*>(smem + L.scal) + e * 4;
      long r = r0 + e;
      float pxv = sc[0], pX = sc[1];
      tv_out[r * 3] = -dptr[0] * pxv;
      tv_out[r * 3 + 1] = -dptr[1] * pxv;
      tv_out[r * 3 + 2] = -dptr[2] * pxv;
      tx_out[r * 3] = dptr[0] * pX;
      tx_out[r * 3 + 1] = dptr[1] * pX;
      tx_out[r * 3 + 2] = dptr[2] * pX;
      if (TRAIN) {
        p2_out[r * 2] = pxv;
        p2_out[r * 2 + 1] = pX;
      }
    }
  }
}

__global__ __launch_bounds__(THREADS, 4) void fused_virtual_bwd(
    const float* __restrict__ coord, const float* __restrict__ vcoord,
    const long* __restrict__ batch,
    const bf16* __restrict__ dvmsg,    // [R,64] cotangent of vmsg
    const float* __restrict__ dtv,     // [R,3]
    const float* __restrict__ dtx,     // [R,3]
    const bf16* __restrict__ z1_in, const bf16* __restrict__ z2_in,
    const bf16* __restrict__ zxv_in, const bf16* __restrict__ zX_in,
    const float* __restrict__ p2_in,   // [R,2]
    const bf16* __restrict__ w1tp,     // [K_OUT][64]
    const bf16* __restrict__ w2t, const bf16* __restrict__ wxvt,
    const bf16* __restrict__ wXt,
    const float* __restrict__ wxvv, const float* __restrict__ wXv,
    bf16* __restrict__ dz1_out, bf16* __restrict__ dz2_out,
    bf16* __restrict__ dzxv_out, bf16* __restrict__ dzX_out,  // [R,64]
    bf16* __restrict__ dh_out,         // [R,64]
    bf16* __restrict__ dvf_out,        // [R,64]
    float* __restrict__ dgram_out,     // [R,CMAX]
    float* __restrict__ dvd_out,       // [R,3]
    float* __restrict__ dp2_out,       // [R,2]
    float* __restrict__ gb_out,  // [6H]: gb1|gb2|gbxv|gbX|gwxvv|gwXv
    long n_rows, int cdim, int k_in) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr VSmemB LB = vsmem_layout_bwd();
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  float* biases = reinterpret_cast<float*>(smem + LB.bias);
  float* gbacc = reinterpret_cast<float*>(smem + LB.gbacc);
  for (int i = tid; i < H; i += THREADS) {
    biases[i] = wxvv[i];
    biases[H + i] = wXv[i];
    for (int kacc = 0; kacc < 6; ++kacc) gbacc[kacc * H + i] = 0.f;
  }
  __syncthreads();  // biases/accumulators initialized by wave 0

  // Wave-per-subtile execution (round 2): no intra-tile barriers — every
  // phase touches only its own wave's 16 rows of each region.
  for (long tile = blockIdx.x; tile * TILE < n_rows; tile += gridDim.x) {
    long r0 = tile * TILE;
    int nrow = (int)((n_rows - r0 < (long)TILE) ? (n_rows - r0)
                                                : (long)TILE);

    // per-row: vdiff recompute, dpxv/dpX, initial dvd (wave-local)
    if (lane < 16) {
      int e = wave * 16 + lane;
      float* dptr = reinterpret_cast<float*>(smem + LB.diff) + e * 4;
      float* sc = reinterpret_cast<float*>(smem + LB.scal) + e * 4;
      float dpxv = 0, dpX = 0;
      if (e < nrow) {
        long r = r0 + e;
        long n = r / cdim, c = r - n * cdim;
        long b = batch[n];
        const float* X = vcoord + (b * cdim + c) * 3;
        float dx = X[0] - coord[n * 3];
        float dy = X[1] - coord[n * 3 + 1];
        float dz = X[2] - coord[n * 3 + 2];
        dptr[0] = dx; dptr[1] = dy; dptr[2] = dz;
        dptr[3] = sqrtf(dx * dx + dy * dy + dz * dz);
        dpxv = -(dtv[r * 3] * dx + dtv[r * 3 + 1] * dy
                 + dtv[r * 3 + 2] * dz);
        dpX = dtx[r * 3] * dx + dtx[r * 3 + 1] * dy + dtx[r * 3 + 2] * dz;
        dp2_out[r * 2] = dpxv;
        dp2_out[r * 2 + 1] = dpX;
      }
      sc[0] = dpxv;
      sc[1] = dpX;
    }

    // dzxv = dpxv * wxv o silu'(zxv); dzX likewise.
    // zxv/zX are staged COALESCED through the (currently free) za and
    // sa regions (wave-local rows).
    for (int idx = lane; idx < 16 * 8; idx += 64) {
      int e = wave * 16 + idx / 8;
      int c8 = (idx % 8) * 8;
      bf16x8 v = {}, w = {};
      if (e < nrow) {
        v = g8(zxv_in + (r0 + e) * H + c8);
        w = g8(zX_in + (r0 + e) * H + c8);
      }
      *reinterpret_cast<bf16x8*>(smem + LB.za + (e * H_STRIDE + c8) * 2) = v;
      *reinterpret_cast<bf16x8*>(smem + LB.sa
                                 + (e * H_STRIDE + c8) * 2) = w;
    }
    {
      int cc = tid & 63;
      int estart = (tid >> 6) * 16;
      float wv0 = biases[cc], wv1 = biases[H + cc];
      __bf16* zc = reinterpret_cast<__bf16*>(smem + LB.zc);
      // dzX overwrites the zX staging in place (same slot, same thread)
      __bf16* zd = reinterpret_cast<__bf16*>(smem + LB.sa);
      const __bf16* zxs = reinterpret_cast<const __bf16*>(smem + LB.za);
      const __bf16* zXs = zd;
      const float* sc = reinterpret_cast<const float*>(smem + LB.scal);
      float abxv = 0.f, abX = 0.f, awxvv = 0.f, awXv = 0.f;
      for (int e = estart; e < estart + 16; ++e) {
        bool ok = e < nrow;
        float zx = (float)zxs[e * H_STRIDE + cc];
        float zX_ = (float)zXs[e * H_STRIDE + cc];
        float dpxv = sc[e * 4], dpX = sc[e * 4 + 1];
        float dxv = dpxv * wv0 * dsilu_(zx);
        float dX = dpX * wv1 * dsilu_(zX_);
        zc[e * H_STRIDE + cc] = (__bf16)dxv;
        zd[e * H_STRIDE + cc] = (__bf16)dX;
        abxv += dxv;
        abX += dX;
        if (ok) {
          awxvv += silu_(zx) * dpxv;
          awXv += silu_(zX_) * dpX;
        }
      }
      atomicAdd(&gbacc[2 * H + cc], abxv);
      atomicAdd(&gbacc[3 * H + cc], abX);
      atomicAdd(&gbacc[4 * H + cc], awxvv);
      atomicAdd(&gbacc[5 * H + cc], awXv);
    }
    TILE_TO_GLOBAL(LB.zc, dzxv_out, false);
    TILE_TO_GLOBAL(LB.sa, dzX_out, false);

    // dvmsg_tot = dvmsg + dzxv@Wxv + dzX@WX; dz2 = dvmsg_tot o silu'(z2)
    {
      f32x4 acc[4] = {};
      mm_g<2, false>(smem, LB.zc + wave * 16 * H_STRIDE * 2, H_STRIDE * 2,
                     opaque(wxvt), H, lane, acc);
      mm_g<2, false>(smem, LB.sa + wave * 16 * H_STRIDE * 2,
                     H_STRIDE * 2, opaque(wXt), H, lane, acc);
      // sa consumed: re-stage za/sa with dvmsg and z2 (wave-local rows)
      for (int idx = lane; idx < 16 * 8; idx += 64) {
        int e = wave * 16 + idx / 8;
        int c8 = (idx % 8) * 8;
        bf16x8 v = {}, w = {};
        if (e < nrow) {
          v = g8(dvmsg + (r0 + e) * H + c8);
          w = g8(z2_in + (r0 + e) * H + c8);
        }
        *reinterpret_cast<bf16x8*>(smem + LB.za + (e * H_STRIDE + c8) * 2) = v;
        *reinterpret_cast<bf16x8*>(smem + LB.sa
                                   + (e * H_STRIDE + c8) * 2) = w;
      }
      __bf16* zb = reinterpret_cast<__bf16*>(smem + LB.zb);
      const __bf16* ups = reinterpret_cast<const __bf16*>(smem + LB.za);
      const __bf16* z2s = reinterpret_cast<const __bf16*>(smem + LB.sa);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        int cc = nt * 16 + (lane & 15);
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          int e = wave * 16 + (lane >> 4) * 4 + rr;
          float up = (float)ups[e * H_STRIDE + cc];
          float z2v = (float)z2s[e * H_STRIDE + cc];
          zb[e * H_STRIDE + cc] =
              (__bf16)((acc[nt][rr] + up) * dsilu_(z2v));
        }
      }
    }
    TILE_TO_GLOBAL(LB.zb, dz2_out, false);
    {
      int cc = tid & 63;
      int estart = (tid >> 6) * 16;
      const __bf16* zb = reinterpret_cast<const __bf16*>(smem + LB.zb);
      float ab = 0.f;
      for (int e = estart; e < estart + 16; ++e)
        ab += (float)zb[e * H_STRIDE + cc];
      atomicAdd(&gbacc[H + cc], ab);
    }

    // dz1 = (dz2 @ W2) o silu'(z1); z1 staged into za coalesced, each
    // slot read once then overwritten in place by its dz1 value
    {
      for (int idx = lane; idx < 16 * 8; idx += 64) {
        int e = wave * 16 + idx / 8;
        int c8 = (idx % 8) * 8;
        bf16x8 v = {};
        if (e < nrow) v = g8(z1_in + (r0 + e) * H + c8);
        *reinterpret_cast<bf16x8*>(smem + LB.za + (e * H_STRIDE + c8) * 2) = v;
      }
      f32x4 acc[4] = {};
      mm_g<2, false>(smem, LB.zb + wave * 16 * H_STRIDE * 2, H_STRIDE * 2,
                     opaque(w2t), H, lane, acc);
      __bf16* za = reinterpret_cast<__bf16*>(smem + LB.za);
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        int cc = nt * 16 + (lane & 15);
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          int e = wave * 16 + (lane >> 4) * 4 + rr;
          float z1v = (float)za[e * H_STRIDE + cc];
          za[e * H_STRIDE + cc] = (__bf16)(acc[nt][rr] * dsilu_(z1v));
        }
      }
    }
    TILE_TO_GLOBAL(LB.za, dz1_out, false);
    {
      int cc = tid & 63;
      int estart = (tid >> 6) * 16;
      const __bf16* za = reinterpret_cast<const __bf16*>(smem + LB.za);
      float ab = 0.f;
      for (int e = estart; e < estart + 16; ++e)
        ab += (float)za[e * H_STRIDE + cc];
      atomicAdd(&gbacc[cc], ab);
    }

    // dvin = dz1 @ W1, streamed through the H_STRIDE-wide sa window in
    // three 48-column passes (a K_STRIDE-wide tile would push the block
    // past the 40 KB that fits 4 blocks/CU). Logical column cc lives at
    // physical cc - pass*48; each pass's dh/dvf columns are stored before
    // the next pass overwrites the window.
#pragma unroll
    for (int pass = 0; pass < 3; ++pass) {
      const bf16* w1tp_ = opaque(w1tp);
      f32x4 acc[3] = {};
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        int k = kk * 32 + (lane >> 4) * 8;
        bf16x8 a = lds8(smem, LB.za + ((wave * 16 + (lane & 15)) * H_STRIDE
                                      + k) * 2);
#pragma unroll
        for (int nt = 0; nt < 3; ++nt) {
          int gc = (pass * 3 + nt) * 16 + (lane & 15);
          bf16x8 b = g8(w1tp_ + gc * H + k);
          acc[nt] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
        }
      }
      __bf16* dvin = reinterpret_cast<__bf16*>(smem + LB.sa);
#pragma unroll
      for (int nt = 0; nt < 3; ++nt) {
        int cc = (pass * 3 + nt) * 16 + (lane & 15);
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          int e = wave * 16 + (lane >> 4) * 4 + rr;
          dvin[e * H_STRIDE + cc - pass * 48] = (__bf16)acc[nt][rr];
          if (cc == 2 * H) {
            reinterpret_cast<float*>(smem + LB.scal)[e * 4 + 2] =
                acc[nt][rr];  // d(vrad)
          }
        }
      }
      // this pass's dh/dvf columns -> global (48-col windows stay
      // 8-aligned; cols >= 2H are the vrad/dgram tail handled below;
      // wave-local rows, so the passes need no barriers)
      for (int idx = lane; idx < 16 * 6; idx += 64) {
        int e = wave * 16 + idx / 6;
        if (e >= nrow) continue;
        int c8 = pass * 48 + (idx % 6) * 8;
        if (c8 >= 2 * H) continue;
        bf16x8 v = lds8(smem, LB.sa + (e * H_STRIDE + c8 - pass * 48) * 2);
        if (c8 < H)
          *reinterpret_cast<bf16x8*>(dh_out + (r0 + e) * H + c8) = v;
        else
          *reinterpret_cast<bf16x8*>(dvf_out + (r0 + e) * H + c8 - H) = v;
      }
    }

    // dgram (logical cols 2H+1.. live in the pass-2 window at physical
    // offset -96) + dvd (per-row)
    if (lane < 16 && wave * 16 + lane < nrow) {
      int e = wave * 16 + lane;
      long r = r0 + e;
      const __bf16* dvin = reinterpret_cast<const __bf16*>(
          smem + LB.sa) + e * H_STRIDE;
      for (int j = 0; j < cdim; ++j)
        dgram_out[r * CMAX + j] = (float)dvin[2 * H + 1 + j - 96];
      for (int j = cdim; j < CMAX; ++j) dgram_out[r * CMAX + j] = 0.f;
      const float* dptr = reinterpret_cast<const float*>(
          smem + LB.diff) + e * 4;
      const float* sc = reinterpret_cast<const float*>(
          smem + LB.scal) + e * 4;
      float p_xv = p2_in[r * 2], p_x = p2_in[r * 2 + 1];
      float drad = sc[2];
      float inv = dptr[3] > 0.f ? drad / dptr[3] : 0.f;
      // dvd = -pxv*dtv + pX*dtx + (vdiff/|vdiff|)*d(vrad)
      dvd_out[r * 3] = -p_xv * dtv[r * 3] + p_x * dtx[r * 3]
                       + dptr[0] * inv;
      dvd_out[r * 3 + 1] = -p_xv * dtv[r * 3 + 1] + p_x * dtx[r * 3 + 1]
                           + dptr[1] * inv;
      dvd_out[r * 3 + 2] = -p_xv * dtv[r * 3 + 2] + p_x * dtx[r * 3 + 2]
                           + dptr[2] * inv;
    }
  }
  __syncthreads();
  for (int c = tid; c < 6 * H; c += THREADS)
    atomicAdd(&gb_out[c], gbacc[c]);
}

}  // namespace

std::vector<torch::Tensor> fused_virtual_forward(
    torch::Tensor h, torch::Tensor coord, torch::Tensor vcoord,
    torch::Tensor vfeat, torch::Tensor gram, torch::Tensor batch,
    torch::Tensor w1, torch::Tensor b1, torch::Tensor w2, torch::Tensor b2,
    torch::Tensor wxv, torch::Tensor bxv, torch::Tensor wxvv,
    torch::Tensor wX, torch::Tensor bX, torch::Tensor wXv, bool train,
    std::vector<torch::Tensor> prepped) {
  TORCH_CHECK(h.is_cuda() && h.scalar_type() == torch::kBFloat16,
              "h must be CUDA bf16");
  TORCH_CHECK(h.size(1) == H, "fused virtual kernel requires hidden_nf=64");
  long n = h.size(0);
  int cdim = (int)vcoord.size(1);
  TORCH_CHECK(cdim <= CMAX, "virtual_channels <= 8");
  int k_in = 2 * H + 1 + cdim;
  long rows = n * cdim;
  auto bopt = h.options();
  auto fopt = coord.options().dtype(torch::kFloat);
  auto vmsg = torch::empty({rows, (long)H}, bopt);
  auto tv = torch::empty({rows, 3}, fopt);
  auto tx = torch::empty({rows, 3}, fopt);
  auto mk = [&](long c, bool f32 = false) {
    return train ? torch::empty({rows, c}, f32 ? fopt : bopt)
                 : torch::empty({0, c}, f32 ? fopt : bopt);
  };
  auto vin = mk(K_OUT);
  auto z1 = mk(H), z2 = mk(H), zxv = mk(H), zX = mk(H);
  auto p2 = mk(2, true);
  if (rows == 0) return {vmsg, tv, tx, vin, z1, z2, zxv, zX, p2};
  auto stream = at::hip::getCurrentHIPStream();
  constexpr VSmem L = vsmem_layout();
  long tiles = (rows + TILE - 1) / TILE;
  int blocks = (int)std::min<long>(tiles, 16384);
  torch::Tensor w1p, w2c, wxvc, wXc, b1c, b2c, bxvc, bXc, wxvvc, wXvc;
  if (!prepped.empty()) {
    TORCH_CHECK(prepped.size() == 10, "virtual fwd prepped wants 10");
    w1p = prepped[0]; w2c = prepped[1]; wxvc = prepped[2]; wXc = prepped[3];
    b1c = prepped[4]; b2c = prepped[5]; bxvc = prepped[6]; bXc = prepped[7];
    wxvvc = prepped[8]; wXvc = prepped[9];
    TORCH_CHECK(w1p.size(1) == K_PAD, "prepped w1p must be row-padded");
  } else {
    w1p = torch::constant_pad_nd(w1.contiguous(), {0, K_PAD - k_in});
    w2c = w2.contiguous(); wxvc = wxv.contiguous(); wXc = wX.contiguous();
    b1c = b1.contiguous().to(torch::kFloat);
    b2c = b2.contiguous().to(torch::kFloat);
    bxvc = bxv.contiguous().to(torch::kFloat);
    bXc = bX.contiguous().to(torch::kFloat);
    wxvvc = wxvv.contiguous().to(torch::kFloat);
    wXvc = wXv.contiguous().to(torch::kFloat);
  }
  auto hc = h.contiguous();
  auto cc_ = coord.contiguous().to(torch::kFloat);
  auto vc = vcoord.contiguous().to(torch::kFloat);
  auto vf = vfeat.contiguous();
  auto gr = gram.contiguous().to(torch::kFloat);
  auto bp = batch.contiguous();

#define ARGS                                                                 \
  reinterpret_cast<const bf16*>(hc.data_ptr()), cc_.data_ptr<float>(),       \
      vc.data_ptr<float>(), reinterpret_cast<const bf16*>(vf.data_ptr()),    \
      gr.data_ptr<float>(), bp.data_ptr<long>(),                             \
      reinterpret_cast<const bf16*>(w1p.data_ptr()),                         \
      reinterpret_cast<const bf16*>(w2c.data_ptr()),                         \
      reinterpret_cast<const bf16*>(wxvc.data_ptr()),                        \
      reinterpret_cast<const bf16*>(wXc.data_ptr()),                         \
      b1c.data_ptr<float>(), b2c.data_ptr<float>(), bxvc.data_ptr<float>(),  \
      bXc.data_ptr<float>(), wxvvc.data_ptr<float>(),                        \
      wXvc.data_ptr<float>(), reinterpret_cast<bf16*>(vmsg.data_ptr()),      \
      tv.data_ptr<float>(), tx.data_ptr<float>(),                            \
      reinterpret_cast<bf16*>(vin.data_ptr()),                               \
      reinterpret_cast<bf16*>(z1.data_ptr()),                                \
      reinterpret_cast<bf16*>(z2.data_ptr()),                                \
      reinterpret_cast<bf16*>(zxv.data_ptr()),                               \
      reinterpret_cast<bf16*>(zX.data_ptr()), p2.data_ptr<float>(), rows,    \
      cdim, k_in

  if (train) {
    fused_virtual_fwd<true><<<blocks, THREADS, L.zd, stream>>>(ARGS);
  } else {
    fused_virtual_fwd<false><<<blocks, THREADS, L.zd, stream>>>(ARGS);
  }
#undef ARGS
  return {vmsg, tv, tx, vin, z1, z2, zxv, zX, p2};
}

std::vector<torch::Tensor> fused_virtual_backward(
    torch::Tensor coord, torch::Tensor vcoord, torch::Tensor batch,
    torch::Tensor dvmsg, torch::Tensor dtv, torch::Tensor dtx,
    torch::Tensor z1, torch::Tensor z2, torch::Tensor zxv, torch::Tensor zX,
    torch::Tensor p2, torch::Tensor w1, torch::Tensor w2, torch::Tensor wxv,
    torch::Tensor wX, torch::Tensor wxvv, torch::Tensor wXv,
    std::vector<torch::Tensor> prepped) {
  long rows = z1.size(0);
  int cdim = (int)vcoord.size(1);
  int k_in = 2 * H + 1 + cdim;
  auto bopt = z1.options();
  auto fopt = coord.options().dtype(torch::kFloat);
  auto dz1 = torch::empty({rows, (long)H}, bopt);
  auto dz2 = torch::empty({rows, (long)H}, bopt);
  auto dzxv = torch::empty({rows, (long)H}, bopt);
  auto dzX = torch::empty({rows, (long)H}, bopt);
  auto dh = torch::empty({rows, (long)H}, bopt);
  auto dvf = torch::empty({rows, (long)H}, bopt);
  auto dgram = torch::empty({rows, (long)CMAX}, fopt);
  auto dvd = torch::empty({rows, 3}, fopt);
  auto dp2 = torch::empty({rows, 2}, fopt);
  auto gb = torch::zeros({6 * (long)H}, fopt);
  if (rows == 0)
    return {dz1, dz2, dzxv, dzX, dh, dvf, dgram, dvd, dp2, gb};
  auto stream = at::hip::getCurrentHIPStream();
  constexpr VSmemB L = vsmem_layout_bwd();
  long tiles = (rows + TILE - 1) / TILE;
  int blocks = (int)std::min<long>(tiles, 16384);
  torch::Tensor w1tp, w2tc, wxvtc, wXtc, wxvvc, wXvc;
  if (!prepped.empty()) {
    TORCH_CHECK(prepped.size() == 6, "virtual bwd prepped wants 6");
    w1tp = prepped[0]; w2tc = prepped[1]; wxvtc = prepped[2];
    wXtc = prepped[3]; wxvvc = prepped[4]; wXvc = prepped[5];
    TORCH_CHECK(w1tp.size(0) == K_OUT, "prepped w1tp must be row-padded");
  } else {
    auto w1c = w1.contiguous();
    w1tp = torch::constant_pad_nd(w1c.t().contiguous(),
                                  {0, 0, 0, K_OUT - k_in});
    w2tc = w2.contiguous().t().contiguous();
    wxvtc = wxv.contiguous().t().contiguous();
    wXtc = wX.contiguous().t().contiguous();
    wxvvc = wxvv.contiguous().to(torch::kFloat);
    wXvc = wXv.contiguous().to(torch::kFloat);
  }
  auto cc_ = coord.contiguous().to(torch::kFloat);
  auto vc = vcoord.contiguous().to(torch::kFloat);
  fused_virtual_bwd<<<blocks, THREADS, L.total, stream>>>(
      cc_.data_ptr<float>(), vc.data_ptr<float>(),
      batch.contiguous().data_ptr<long>(),
      reinterpret_cast<const bf16*>(dvmsg.contiguous().data_ptr()),
      dtv.contiguous().data_ptr<float>(),
      dtx.contiguous().data_ptr<float>(),
      reinterpret_cast<const bf16*>(z1.data_ptr()),
      reinterpret_cast<const bf16*>(z2.data_ptr()),
      reinterpret_cast<const bf16*>(zxv.data_ptr()),
      reinterpret_cast<const bf16*>(zX.data_ptr()), p2.data_ptr<float>(),
      reinterpret_cast<const bf16*>(w1tp.data_ptr()),
      reinterpret_cast<const bf16*>(w2tc.data_ptr()),
      reinterpret_cast<const bf16*>(wxvtc.data_ptr()),
      reinterpret_cast<const bf16*>(wXtc.data_ptr()),
      wxvvc.data_ptr<float>(), wXvc.data_ptr<float>(),
      reinterpret_cast<bf16*>(dz1.data_ptr()),
      reinterpret_cast<bf16*>(dz2.data_ptr()),
      reinterpret_cast<bf16*>(dzxv.data_ptr()),
      reinterpret_cast<bf16*>(dzX.data_ptr()),
      reinterpret_cast<bf16*>(dh.data_ptr()),
      reinterpret_cast<bf16*>(dvf.data_ptr()), dgram.data_ptr<float>(),
      dvd.data_ptr<float>(), dp2.data_ptr<float>(), gb.data_ptr<float>(),
      rows, cdim, k_in);
  return {dz1, dz2, dzxv, dzX, dh, dvf, dgram, dvd, dp2, gb};
}
