// CDNA4 (gfx950) kernels for SAE-ensemble training, fp32 end to end.
//
// These implement the reference's implicit kernel inventory (SURVEY.md §2.4
// K1-K6: encoder GEMM -> ReLU -> decoder GEMM -> MSE+L1 forward, analytic
// backward through the in-forward decoder row renormalization, fused Adam)
// as hand-written MFMA kernels.  fp32 is the reference's training dtype
// (BASELINE.md); we use the exact-f32 matrix instruction
// v_mfma_f32_32x32x2_f32 (157 TF peak, bitwise == an fmaf chain).
//
// GEMM core structure (tuned from rocprof on MI355X):
//  * 128x128 output tile, 8 waves (512 threads); each wave owns a 32x64
//    quadrant as 2 v_mfma_f32_32x32x2_f32 accumulators (32 AGPRs).
//  * K-tile depth TBK is a template parameter:
//      TBK=32, __launch_bounds__(512,4): 64-66 KB LDS, 2 blocks/CU.
//      TBK=16, __launch_bounds__(512,6): 32-33 KB LDS, <=80 VGPRs,
//        3 blocks/CU — more co-resident blocks to hide barrier skew
//        (SQ_WAIT_ANY was 25-33% of wave cycles at 2 blocks/CU).
//    Both are compiled; the launcher picks at runtime (ops.kconfig).
//  * ping-pong LDS double buffer + register prefetch (guide T14/G15): ONE
//    barrier per K-step; the NEXT K-tile's global loads issue before the
//    MFMA phase of the current tile.
//  * LDS staging is conflict-free: transposed tiles use stride BM+1 (b32
//    writes); direct tiles use stride BM with float4 (b128) writes.
//  * optional s_setprio(1) on the second-dispatched wave half (runtime
//    `prio` flag; MI355X_MICROARCH "Two waves per SIMD" item 4).
//  * XCD-aware bijective block swizzle (guide T1): consecutive remapped
//    blocks land on one XCD and share operand panels in its private L2.
//
// Kernels (each a template over TBK, MINW):
//   k_enc_fwd_t  : c = relu(x @ Wenc^T + b)    (+ L1 partial, fired counts)
//   k_enc_fwd2_t : same with pre-transposed operands (all-direct staging)
//   k_dec_fwd_t  : r = c @ Wdec_hat - x        (+ MSE partial)
//   k_gc_t/k_gc2_t: gpre = (c>0) .* (gs * r @ Wdec_hat^T + l1/B) (+ bias-grad)
//   k_grad_w_t   : gw = beta*gw + alpha * P^T @ Q   (K = batch contraction)
//   k_row_norms, k_project_adam (gradient of w/max(||w||,eps)), k_bias_adam,
//   k_transpose_scale, k_colsum (HBM-rate column sums / L1 reductions),
//   k_topk_select (exact per-row radix top-k), k_resample (K14 on-device),
//   k_lista_bwd_elem (one-pass LISTA backward elementwise chain)

#include <hip/hip_runtime.h>

#define WAVE 64
#define BM 128
#define BN 128
#define BMP (BM + 1)  // padded LDS stride for transposed staging
#define NTHREADS 512
#define NXCD 8

typedef float f32x16 __attribute__((ext_vector_type(16)));

// C/D fragment mapping for 32x32 MFMA (guide §3): reg r, lane l ->
//   row = (r&3) + 8*(r>>2) + 4*(l>>5), col = l&31
__device__ __forceinline__ int acc_row(int r, int lane) {
  return (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
}

// Bijective XCD-contiguous remap of a linear block id (guide T1; bijective
// variant for nwg % 8 != 0).
__device__ __forceinline__ long xcd_swizzle(long id, long nwg) {
  long q = nwg / NXCD, r = nwg % NXCD;
  long xcd = id % NXCD, pos = id / NXCD;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
}

// Decompose blockIdx (x = col tile, y = row tile, z = model) after swizzling
// the (x, y) plane per model so that co-XCD blocks share operand panels.
__device__ __forceinline__ void tile_coords(int& tx, int& ty) {
  long nwg = (long)gridDim.x * gridDim.y;
  long id = (long)blockIdx.x + (long)gridDim.x * blockIdx.y;
  long s = xcd_swizzle(id, nwg);
  tx = (int)(s % gridDim.x);
  ty = (int)(s / gridDim.x);
}

// Optional static priority for the second-dispatched wave half: the younger
// waves lose VALU arbitration on every segment; one setprio before the main
// loop, no per-segment flips.
__device__ __forceinline__ void maybe_prio(int prio) {
  if (prio && threadIdx.x >= NTHREADS / 2) __builtin_amdgcn_s_setprio(1);
}

// ---------------------------------------------------------------------------
// staging: load (global -> regs) and write (regs -> LDS) split so the loads
// overlap the previous tile's MFMA phase.  TBK/16 float4 passes per thread.
// ---------------------------------------------------------------------------

// Transposed tile: lds[k][i] = src[i0+i][k0+k] (* scale[i0+i]), i<TROWS, k<TBK.
template <int TBK, int TROWS = BM>
struct TStage {
  float4 v[TBK * TROWS / 2048];
  float sc[TBK * TROWS / 2048];
};

template <int TBK, int TROWS = BM>
__device__ __forceinline__ void stage_T_load(const float* __restrict__ src, long ld,
                                             int i0, int k0, int n_rows, int n_k,
                                             const float* __restrict__ scale,
                                             TStage<TBK, TROWS>& st) {
  const int t = threadIdx.x;
  constexpr int TPR = TBK / 4;                 // threads covering one row's K
  constexpr int RPP = NTHREADS / TPR;          // rows per pass
#pragma unroll
  for (int p = 0; p < TBK * TROWS / 2048; ++p) {
    int i = p * RPP + t / TPR;
    int kc = (t % TPR) * 4;
    int gi = i0 + i;
    float4 v = make_float4(0.f, 0.f, 0.f, 0.f);
    float sc = 1.f;
    if (gi < n_rows) {
      const float* row = src + (long)gi * ld + k0 + kc;
      if (k0 + kc + 3 < n_k) {
        v = *reinterpret_cast<const float4*>(row);
      } else {
#pragma unroll
        for (int s = 0; s < 4; ++s)
          if (k0 + kc + s < n_k) ((float*)&v)[s] = row[s];
      }
      if (scale) sc = scale[gi];
    }
    st.v[p] = v;
    st.sc[p] = sc;
  }
}

template <int TBK, int TROWS = BM>
__device__ __forceinline__ void stage_T_write(const TStage<TBK, TROWS>& st, float* __restrict__ lds,
                                              bool scaled) {
  const int t = threadIdx.x;
  constexpr int TPR = TBK / 4;
  constexpr int RPP = NTHREADS / TPR;
#pragma unroll
  for (int p = 0; p < TBK * TROWS / 2048; ++p) {
    int i = p * RPP + t / TPR;
    int kc = (t % TPR) * 4;
    float4 v = st.v[p];
    if (scaled) {
      v.x *= st.sc[p]; v.y *= st.sc[p]; v.z *= st.sc[p]; v.w *= st.sc[p];
    }
#pragma unroll
    for (int s = 0; s < 4; ++s) lds[(kc + s) * (TROWS + 1) + i] = ((float*)&v)[s];
  }
}

// Direct tile: lds[k][j] = src[k0+k][j0+j] (* scale[k0+k]), k<TBK, j<TCOLS.
template <int TBK, int TCOLS = BM>
struct DStage {
  float4 v[TBK * TCOLS / 2048];
  float sc[TBK * TCOLS / 2048];
};

template <int TBK, int TCOLS = BM>
__device__ __forceinline__ void stage_D_load(const float* __restrict__ src, long ld,
                                             int k0, int j0, int n_k, int n_cols,
                                             const float* __restrict__ scale,
                                             DStage<TBK, TCOLS>& st) {
  const int t = threadIdx.x;
  constexpr int TPC = TCOLS / 4;               // threads covering one k's cols
  constexpr int KPP = NTHREADS / TPC;          // k rows per pass
#pragma unroll
  for (int p = 0; p < TBK * TCOLS / 2048; ++p) {
    int k = p * KPP + t / TPC;
    int j = (t % TPC) * 4;
    int gk = k0 + k;
    float4 v = make_float4(0.f, 0.f, 0.f, 0.f);
    float sc = 1.f;
    if (gk < n_k) {
      const float* row = src + (long)gk * ld + j0 + j;
      if (j0 + j + 3 < n_cols) {
        v = *reinterpret_cast<const float4*>(row);
      } else {
#pragma unroll
        for (int s = 0; s < 4; ++s)
          if (j0 + j + s < n_cols) ((float*)&v)[s] = row[s];
      }
      if (scale) sc = scale[gk];
    }
    st.v[p] = v;
    st.sc[p] = sc;
  }
}

template <int TBK, int TCOLS = BM>
__device__ __forceinline__ void stage_D_write(const DStage<TBK, TCOLS>& st, float* __restrict__ lds,
                                              bool scaled) {
  const int t = threadIdx.x;
  constexpr int TPC = TCOLS / 4;
  constexpr int KPP = NTHREADS / TPC;
#pragma unroll
  for (int p = 0; p < TBK * TCOLS / 2048; ++p) {
    int k = p * KPP + t / TPC;
    int j = (t % TPC) * 4;
    float4 v = st.v[p];
    if (scaled) {
      v.x *= st.sc[p]; v.y *= st.sc[p]; v.z *= st.sc[p]; v.w *= st.sc[p];
    }
    *reinterpret_cast<float4*>(&lds[k * TCOLS + j]) = v;
  }
}

// ---------------------------------------------------------------------------
// MFMA phase: 8 waves; wave w covers rows [(w&3)*32, +32), cols [(w>>2)*64, +64)
// ---------------------------------------------------------------------------
template <int TBK, int NACC, int ASTRIDE, int BSTRIDE>
__device__ __forceinline__ void mfma_tile(const float* __restrict__ As,
                                          const float* __restrict__ Bs,
                                          f32x16 acc[NACC]) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int wr = (wave & 3) * 32;
  const int wc = (wave >> 2) * (NACC * 32);
  const int l31 = lane & 31;
  const int h = lane >> 5;

#pragma unroll
  for (int kk = 0; kk < TBK; kk += 2) {
    float a0 = As[(kk + h) * ASTRIDE + wr + l31];
#pragma unroll
    for (int j = 0; j < NACC; ++j) {
      float b = Bs[(kk + h) * BSTRIDE + wc + j * 32 + l31];
      acc[j] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b, acc[j], 0, 0, 0);
    }
  }
}

template <int NACC>
__device__ __forceinline__ void zero_acc(f32x16 acc[NACC]) {
#pragma unroll
  for (int j = 0; j < NACC; ++j)
#pragma unroll
    for (int r = 0; r < 16; ++r) acc[j][r] = 0.f;
}

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

// epilogue lane geometry shared by all GEMM kernels
struct EpiGeom {
  int lane, wave, wr, wc, l31;
};
template <int NACC = 2>
__device__ __forceinline__ EpiGeom epi_geom() {
  EpiGeom g;
  g.lane = threadIdx.x & (WAVE - 1);
  g.wave = threadIdx.x / WAVE;
  g.wr = (g.wave & 3) * 32;
  g.wc = (g.wave >> 2) * (NACC * 32);
  g.l31 = g.lane & 31;
  return g;
}

// ---------------------------------------------------------------------------
// k_row_norms
// ---------------------------------------------------------------------------
extern "C" __global__ void k_row_norms(const float* __restrict__ W,
                                       float* __restrict__ norms,
                                       float* __restrict__ inv_norms,
                                       int n_rows_total, int d, float eps) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int row = blockIdx.x * (NTHREADS / WAVE) + wave;
  if (row >= n_rows_total) return;
  const float* w = W + (long)row * d;
  float s = 0.f;
  if (d % (WAVE * 4) == 0) {
    for (int j = lane * 4; j < d; j += WAVE * 4) {
      float4 v = *reinterpret_cast<const float4*>(w + j);
      s += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
    }
  } else {
    for (int j = lane; j < d; j += WAVE) s += w[j] * w[j];
  }
  s = wave_reduce_sum(s);
  if (lane == 0) {
    float nrm = sqrtf(s);
    norms[row] = nrm;
    inv_norms[row] = 1.0f / fmaxf(nrm, eps);
  }
}

// ---------------------------------------------------------------------------
// the shared prefetch K-loop (macro: operands differ per kernel)
// ---------------------------------------------------------------------------
// Ping-pong double buffer + register prefetch: ONE barrier per K-step.
// Iteration t: issue loads of tile t+1, MFMA on buffer `cur` (tile t), write
// tile t+1 into buffer cur^1 (nobody reads it), barrier, flip.  The barrier
// at the end of iteration t also protects the write of tile t+2 into the
// old `cur` (every wave has finished reading it).
#define PREFETCH_LOOP(TBK, K_TOTAL, LOAD_A, LOAD_B, WRITE_A, WRITE_B, ASTR, BSTR) \
  {                                                                            \
    int cur = 0;                                                               \
    LOAD_A(0);                                                                 \
    LOAD_B(0);                                                                 \
    WRITE_A(0);                                                                \
    WRITE_B(0);                                                                \
    __syncthreads();                                                           \
    for (int k0 = (TBK); k0 < (K_TOTAL); k0 += (TBK)) {                        \
      LOAD_A(k0);                                                              \
      LOAD_B(k0);                                                              \
      mfma_tile<TBK, NACC, ASTR, BSTR>(&As[cur][0], &Bs[cur][0], acc);         \
      WRITE_A(cur ^ 1);                                                        \
      WRITE_B(cur ^ 1);                                                        \
      __syncthreads();                                                         \
      cur ^= 1;                                                                \
    }                                                                          \
    mfma_tile<TBK, NACC, ASTR, BSTR>(&As[cur][0], &Bs[cur][0], acc);           \
  }

// The soft-threshold gate of the thresholding SAE (reference
// sae_ensemble.py:256-259): u = (c + gain) / max(a^2, eps),
// g(u) = relu6(60(u-0.9))/6 + relu(u-1), code = g(u) * a^2 (RAW a^2 — the
// reference multiplies by the unclamped square).
#define GATE_EPS 1e-8f
__device__ __forceinline__ float gate_g(float u) {
  return fminf(fmaxf(60.0f * (u - 0.9f), 0.0f), 6.0f) * (1.0f / 6.0f) +
         fmaxf(u - 1.0f, 0.0f);
}
__device__ __forceinline__ float gate_gp(float u) {  // dg/du (0 at the kinks)
  float gp = (u > 0.9f && u < 1.0f) ? 10.0f : 0.0f;
  if (u > 1.0f) gp += 1.0f;
  return gp;
}

// ---------------------------------------------------------------------------
// k_enc_fwd_t
// ---------------------------------------------------------------------------
template <int TBK, int MINW, int TBN = BN>
__global__ __launch_bounds__(NTHREADS, MINW)
void k_enc_fwd_t(const float* __restrict__ x,       // [B, d]
                 const float* __restrict__ Wenc,    // [M, n, d]
                 const float* __restrict__ bias,    // [M, n]
                 const float* __restrict__ inv_norms, // [M, n] or nullptr
                 float* __restrict__ c_out,         // [M, B, n]
                 float* __restrict__ loss_parts,    // [M, 2]
                 float* __restrict__ fired,         // [M, n]
                 int B, int d, int n,
                 int mode,  // 0: bias+relu (+L1/fired); 1: raw scores (TopK);
                            // 2: threshold gate via act_scale/act_gain/u_out
                 int prio,
                 const float* __restrict__ act_scale,  // [M, n] (mode 2)
                 const float* __restrict__ act_gain,   // [M, n] (mode 2)
                 float* __restrict__ u_out,            // [M, B, n] (mode 2/4)
                 const int* __restrict__ dict_sizes, // [M] or nullptr:
                                       // masked sigs zero cols >= dict_sizes[m]
                 long x_mstride,       // 0: x shared [B,d]; else x is [M,B,d]
                 const float* __restrict__ y_in,   // [M, B, n] (modes 4, 5)
                 const float* __restrict__ x_prev, // [M, B, n] (mode 4)
                 float* __restrict__ x_out,        // [M, B, n] (mode 4)
                 const float* __restrict__ mom) {  // [M] momentum (mode 4)
  __shared__ float As[2][TBK * BMP];
  __shared__ float Bs[2][TBK * (TBN + 1)];

  maybe_prio(prio);
  const int m = blockIdx.z;
  int tx, ty;
  tile_coords(tx, ty);
  const int row0 = ty * BM;
  const int col0 = tx * TBN;
  const float* W = Wenc + (long)m * n * d;
  const float* x_m = x + (long)m * x_mstride;
  const float* inv = inv_norms ? inv_norms + (long)m * n : nullptr;
  const bool scaled = inv != nullptr;

  constexpr int NACC = TBN / 64;
  f32x16 acc[NACC];
  zero_acc<NACC>(acc);
  TStage<TBK> sa;
  TStage<TBK, TBN> sb;

#define ENC_LA(K) stage_T_load<TBK>(x_m, d, row0, (K), B, d, nullptr, sa)
#define ENC_LB(K) stage_T_load<TBK, TBN>(W, d, col0, (K), n, d, inv, sb)
#define ENC_WA(BUF) stage_T_write<TBK>(sa, &As[BUF][0], false)
#define ENC_WB(BUF) stage_T_write<TBK, TBN>(sb, &Bs[BUF][0], scaled)
  PREFETCH_LOOP(TBK, d, ENC_LA, ENC_LB, ENC_WA, ENC_WB, BMP, TBN + 1)
#undef ENC_LA
#undef ENC_LB
#undef ENC_WA
#undef ENC_WB

  const EpiGeom g = epi_geom<NACC>();
  float* c_m = c_out + (long)m * B * n;
  const float* bias_m = bias + (long)m * n;
  float* fired_m = fired + (long)m * n;

  float l1_sum = 0.f;
#pragma unroll
  for (int tj = 0; tj < NACC; ++tj) {
    int col = col0 + g.wc + tj * 32 + g.l31;
    bool col_ok = col < n;
    if (mode == 1) {
      // raw scores for TopK selection: no bias, no relu, no side outputs
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int row = row0 + g.wr + acc_row(r, g.lane);
        if (row < B && col_ok) c_m[(long)row * n + col] = acc[tj][r];
      }
      continue;
    }
    if (mode == 2) {
      // thresholding SAE: code = g((c+gain)/max(a^2,eps)) * a^2; u kept for
      // the backward's gate derivative (k_gc_thresh_t)
      float a = col_ok ? act_scale[(long)m * n + col] : 1.f;
      float gn = col_ok ? act_gain[(long)m * n + col] : 0.f;
      float s_raw = a * a;
      float s_inv = 1.0f / fmaxf(s_raw, GATE_EPS);
      float* u_m = u_out + (long)m * B * n;
      float fired_cnt = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int row = row0 + g.wr + acc_row(r, g.lane);
        if (row < B && col_ok) {
          float u = (acc[tj][r] + gn) * s_inv;
          float code = gate_g(u) * s_raw;
          c_m[(long)row * n + col] = code;
          u_m[(long)row * n + col] = u;
          l1_sum += code;
          fired_cnt += (code > 0.f) ? 1.f : 0.f;
        }
      }
      if (col_ok) {
        float other = __shfl_xor(fired_cnt, 32, WAVE);
        float tot = fired_cnt + other;
        if (g.lane < 32 && tot > 0.f) atomicAdd(&fired_m[col], tot);
      }
      continue;
    }
    if (mode == 4 || mode == 5) {
      // LISTA layer epilogues (HipLISTAStep): the GEMM result `acc` is
      //   mode 4:  s = neg_e W_l^T;  r = y - s;  x = shrink(r, theta);
      //            y' = x + m (x - x_prev)   -> u_out=r, x_out=x, c_out=y'
      //   mode 5:  c_out = y_in - acc        (backward g_y = g_r - g_e A^T)
      const float* y_m = y_in + (long)m * (long)B * n;
      float th = 0.f, mm = 0.f;
      const float* xp_m = nullptr;
      float* xo_m = nullptr;
      float* u_m = nullptr;
      if (mode == 4) {
        th = col_ok ? act_scale[(long)m * n + col] : 0.f;
        mm = mom[m];
        xp_m = x_prev + (long)m * (long)B * n;
        xo_m = x_out + (long)m * (long)B * n;
        u_m = u_out + (long)m * (long)B * n;
      }
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int row = row0 + g.wr + acc_row(r, g.lane);
        if (row < B && col_ok) {
          long idx = (long)row * n + col;
          float v = y_m[idx] - acc[tj][r];
          if (mode == 5) {
            c_m[idx] = v;
          } else {
            u_m[idx] = v;  // r_l, kept for the backward's shrink mask
            float a = fabsf(v) - th;
            float xa = (a > 0.f) ? ((v > 0.f) ? a : -a) : 0.f;
            xo_m[idx] = xa;
            c_m[idx] = xa + mm * (xa - xp_m[idx]);
          }
        }
      }
      continue;
    }
    if (mode == 3) {
      // reverse SAE (sae_ensemble.py:447-503): code = pre * [pre + b > 0]
      // (the bias is removed from active features before decoding, so the
      // code can be negative and the L1 partial needs |code|)
      float bj = col_ok ? bias_m[col] : 0.f;
      float fired_cnt = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int row = row0 + g.wr + acc_row(r, g.lane);
        if (row < B && col_ok) {
          float pre = acc[tj][r];
          float v = (pre + bj > 0.f) ? pre : 0.f;
          c_m[(long)row * n + col] = v;
          l1_sum += fabsf(v);
          fired_cnt += (v != 0.f) ? 1.f : 0.f;
        }
      }
      if (col_ok) {
        float other = __shfl_xor(fired_cnt, 32, WAVE);
        float tot = fired_cnt + other;
        if (g.lane < 32 && tot > 0.f) atomicAdd(&fired_m[col], tot);
      }
      continue;
    }
    float bj = col_ok ? bias_m[col] : 0.f;
    // coefficient mask (reference K9): columns >= dict_sizes[m] are dead
    bool live = col_ok && (!dict_sizes || col < dict_sizes[m]);
    float fired_cnt = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int row = row0 + g.wr + acc_row(r, g.lane);
      if (row < B && col_ok) {
        float v = live ? fmaxf(acc[tj][r] + bj, 0.f) : 0.f;
        c_m[(long)row * n + col] = v;
        l1_sum += v;
        fired_cnt += (v > 0.f) ? 1.f : 0.f;
      }
    }
    if (col_ok) {
      float other = __shfl_xor(fired_cnt, 32, WAVE);
      float tot = fired_cnt + other;
      if (g.lane < 32 && tot > 0.f) atomicAdd(&fired_m[col], tot);
    }
  }
  if (mode == 0 || mode == 2 || mode == 3) {
    l1_sum = wave_reduce_sum(l1_sum);
    if (g.lane == 0) atomicAdd(&loss_parts[m * 2 + 1], l1_sum);
  }
}

// ---------------------------------------------------------------------------
// k_dec_fwd_t
// ---------------------------------------------------------------------------
template <int TBK, int MINW, int TBN = BN>
__global__ __launch_bounds__(NTHREADS, MINW)
void k_dec_fwd_t(const float* __restrict__ c,       // [M, B, n]
                 const float* __restrict__ Wdec,    // [M, n, d]
                 const float* __restrict__ inv_norms, // [M, n]
                 const float* __restrict__ x,       // [B, d]
                 float* __restrict__ r_out,         // [M, B, d]
                 float* __restrict__ loss_parts,    // [M, 2]
                 int B, int d, int n, int prio,
                 long x_mstride) {  // 0: shared [B,d]; else [M,B,d]
  __shared__ float As[2][TBK * BMP];
  __shared__ float Bs[2][TBK * TBN];

  maybe_prio(prio);
  const int m = blockIdx.z;
  int tx, ty;
  tile_coords(tx, ty);
  const int row0 = ty * BM;
  const int col0 = tx * TBN;
  const float* c_m = c + (long)m * B * n;
  const float* W = Wdec + (long)m * n * d;
  const float* x_m = x + (long)m * x_mstride;
  const float* inv = inv_norms + (long)m * n;

  constexpr int NACC = TBN / 64;
  f32x16 acc[NACC];
  zero_acc<NACC>(acc);
  TStage<TBK> sa;
  DStage<TBK, TBN> sb;

#define DEC_LA(K) stage_T_load<TBK>(c_m, n, row0, (K), B, n, nullptr, sa)
#define DEC_LB(K) stage_D_load<TBK, TBN>(W, d, (K), col0, n, d, inv, sb)
#define DEC_WA(BUF) stage_T_write<TBK>(sa, &As[BUF][0], false)
#define DEC_WB(BUF) stage_D_write<TBK, TBN>(sb, &Bs[BUF][0], true)
  PREFETCH_LOOP(TBK, n, DEC_LA, DEC_LB, DEC_WA, DEC_WB, BMP, TBN)
#undef DEC_LA
#undef DEC_LB
#undef DEC_WA
#undef DEC_WB

  const EpiGeom g = epi_geom<NACC>();
  float* r_m = r_out + (long)m * B * d;

  float mse_sum = 0.f;
#pragma unroll
  for (int tj = 0; tj < NACC; ++tj) {
    int col = col0 + g.wc + tj * 32 + g.l31;
    bool col_ok = col < d;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int row = row0 + g.wr + acc_row(r, g.lane);
      if (row < B && col_ok) {
        float rv = acc[tj][r] - x_m[(long)row * d + col];
        r_m[(long)row * d + col] = rv;
        mse_sum += rv * rv;
      }
    }
  }
  mse_sum = wave_reduce_sum(mse_sum);
  if (g.lane == 0) atomicAdd(&loss_parts[m * 2 + 0], mse_sum);
}

// ---------------------------------------------------------------------------
// k_gc_t
// ---------------------------------------------------------------------------
template <int TBK, int MINW, int TBN = BN>
__global__ __launch_bounds__(NTHREADS, MINW)
void k_gc_t(const float* __restrict__ r,        // [M, B, d]
            const float* __restrict__ Wdec,     // [M, n, d]
            const float* __restrict__ inv_norms,// [M, n]
            const float* __restrict__ c,        // [M, B, n]
            const float* __restrict__ l1_alpha, // [M]
            float* __restrict__ gpre_out,       // [M, B, n]
            float* __restrict__ g_bias,         // [M, n]
            int B, int d, int n, int prio, int gc_mode) {
  __shared__ float As[2][TBK * BMP];
  __shared__ float Bs[2][TBK * (TBN + 1)];

  maybe_prio(prio);
  const int m = blockIdx.z;
  int tx, ty;
  tile_coords(tx, ty);
  const int row0 = ty * BM;
  const int col0 = tx * TBN;
  const float* r_m = r + (long)m * B * d;
  const float* W = Wdec + (long)m * n * d;
  const float* inv = inv_norms + (long)m * n;
  const float* c_m = c + (long)m * B * n;
  const float gscale = 2.0f / ((float)B * (float)d);
  const float l1_term = l1_alpha[m] / (float)B;

  constexpr int NACC = TBN / 64;
  f32x16 acc[NACC];
  zero_acc<NACC>(acc);
  TStage<TBK> sa;
  TStage<TBK, TBN> sb;

#define GC_LA(K) stage_T_load<TBK>(r_m, d, row0, (K), B, d, nullptr, sa)
#define GC_LB(K) stage_T_load<TBK, TBN>(W, d, col0, (K), n, d, inv, sb)
#define GC_WA(BUF) stage_T_write<TBK>(sa, &As[BUF][0], false)
#define GC_WB(BUF) stage_T_write<TBK, TBN>(sb, &Bs[BUF][0], true)
  PREFETCH_LOOP(TBK, d, GC_LA, GC_LB, GC_WA, GC_WB, BMP, TBN + 1)
#undef GC_LA
#undef GC_LB
#undef GC_WA
#undef GC_WB

  const EpiGeom g = epi_geom<NACC>();
  float* g_m = gpre_out + (long)m * B * n;
  float* gb_m = g_bias + (long)m * n;

#pragma unroll
  for (int tj = 0; tj < NACC; ++tj) {
    int col = col0 + g.wc + tj * 32 + g.l31;
    bool col_ok = col < n;
    float colsum = 0.f;
#pragma unroll
    for (int r_ = 0; r_ < 16; ++r_) {
      int row = row0 + g.wr + acc_row(r_, g.lane);
      if (row < B && col_ok) {
        float cv = c_m[(long)row * n + col];
        float gv;
        if (gc_mode == 1) {
          // reverse SAE: active set is cv != 0, l1 grad is sign(cv), and
          // the bias receives no gradient from the code path
          float sgn = (cv > 0.f) ? 1.f : ((cv < 0.f) ? -1.f : 0.f);
          gv = (cv != 0.f) ? (gscale * acc[tj][r_] + l1_term * sgn) : 0.f;
        } else {
          gv = (cv > 0.f) ? (gscale * acc[tj][r_] + l1_term) : 0.f;
        }
        g_m[(long)row * n + col] = gv;
        colsum += gv;
      }
    }
    if (col_ok && gc_mode == 0) {
      float other = __shfl_xor(colsum, 32, WAVE);
      float tot = colsum + other;
      if (g.lane < 32 && tot != 0.f) atomicAdd(&gb_m[col], tot);
    }
  }
}

// ---------------------------------------------------------------------------
// k_gc_thresh_t: backward of the thresholding SAE's gate (K15).
// Same GEMM as k_gc_t (acc = r @ What^T); epilogue turns dL/dcode into
//   gpre   = dL/dc      = gv * g'(u) * a^2/max(a^2,eps)   (feeds grad_w)
//   g_gain = sum_b dL/dgain = same weight as gpre (column sum)
//   g_scale= sum_b dL/da = gv * 2a * (g(u) - [a^2>eps] g'(u) u a^2/max(..))
// with gv = gscale*acc + l1/B * [code>0].
// ---------------------------------------------------------------------------
template <int TBK, int MINW, int TBN = BN>
__global__ __launch_bounds__(NTHREADS, MINW)
void k_gc_thresh_t(const float* __restrict__ r,        // [M, B, d]
                   const float* __restrict__ Wdec,     // [M, n, d]
                   const float* __restrict__ inv_norms,// [M, n]
                   const float* __restrict__ c,        // [M, B, n] (codes)
                   const float* __restrict__ u,        // [M, B, n]
                   const float* __restrict__ act_scale,// [M, n]
                   const float* __restrict__ l1_alpha, // [M]
                   float* __restrict__ gpre_out,       // [M, B, n]
                   float* __restrict__ g_gain,         // [M, n]
                   float* __restrict__ g_scale,        // [M, n]
                   int B, int d, int n, int prio) {
  __shared__ float As[2][TBK * BMP];
  __shared__ float Bs[2][TBK * (TBN + 1)];

  maybe_prio(prio);
  const int m = blockIdx.z;
  int tx, ty;
  tile_coords(tx, ty);
  const int row0 = ty * BM;
  const int col0 = tx * TBN;
  const float* r_m = r + (long)m * B * d;
  const float* W = Wdec + (long)m * n * d;
  const float* inv = inv_norms + (long)m * n;
  const float* c_m = c + (long)m * B * n;
  const float* u_m = u + (long)m * B * n;
  const float gscale = 2.0f / ((float)B * (float)d);
  const float l1_term = l1_alpha[m] / (float)B;

  constexpr int NACC = TBN / 64;
  f32x16 acc[NACC];
  zero_acc<NACC>(acc);
  TStage<TBK> sa;
  TStage<TBK, TBN> sb;

#define GCT_LA(K) stage_T_load<TBK>(r_m, d, row0, (K), B, d, nullptr, sa)
#define GCT_LB(K) stage_T_load<TBK, TBN>(W, d, col0, (K), n, d, inv, sb)
#define GCT_WA(BUF) stage_T_write<TBK>(sa, &As[BUF][0], false)
#define GCT_WB(BUF) stage_T_write<TBK, TBN>(sb, &Bs[BUF][0], true)
  PREFETCH_LOOP(TBK, d, GCT_LA, GCT_LB, GCT_WA, GCT_WB, BMP, TBN + 1)
#undef GCT_LA
#undef GCT_LB
#undef GCT_WA
#undef GCT_WB

  const EpiGeom g = epi_geom<NACC>();
  float* g_m = gpre_out + (long)m * B * n;
  float* gg_m = g_gain + (long)m * n;
  float* gs_m = g_scale + (long)m * n;

#pragma unroll
  for (int tj = 0; tj < NACC; ++tj) {
    int col = col0 + g.wc + tj * 32 + g.l31;
    bool col_ok = col < n;
    float a = col_ok ? act_scale[(long)m * n + col] : 1.f;
    float s_raw = a * a;
    float s = fmaxf(s_raw, GATE_EPS);
    float clamp_act = (s_raw > GATE_EPS) ? 1.f : 0.f;  // clamp passes grad?
    float f = s_raw / s;  // a^2 / clamp(a^2): ==1 away from the clamp
    float gain_colsum = 0.f, scale_colsum = 0.f;
#pragma unroll
    for (int r_ = 0; r_ < 16; ++r_) {
      int row = row0 + g.wr + acc_row(r_, g.lane);
      if (row < B && col_ok) {
        long idx = (long)row * n + col;
        float code = c_m[idx];
        float uv = u_m[idx];
        float gp = gate_gp(uv);
        float gv = gscale * acc[tj][r_] + ((code > 0.f) ? l1_term : 0.f);
        float gc_ = gv * gp * f;
        g_m[idx] = gc_;
        gain_colsum += gc_;
        scale_colsum += gv * 2.0f * a * (gate_g(uv) - clamp_act * gp * uv * f);
      }
    }
    if (col_ok) {
      float og = __shfl_xor(gain_colsum, 32, WAVE);
      float os = __shfl_xor(scale_colsum, 32, WAVE);
      if (g.lane < 32) {
        if (gain_colsum + og != 0.f) atomicAdd(&gg_m[col], gain_colsum + og);
        if (scale_colsum + os != 0.f) atomicAdd(&gs_m[col], scale_colsum + os);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// k_lista_bwd_elem: one pass over [M, B, n] replacing the LISTA backward's
// elementwise chain:
//   g_x   = (1 + m) g_y + carry_in
//   g_r   = g_x * [|r| > theta]
//   carry = -m g_y                       (becomes next layer's carry_in)
//   g_theta[col] += -g_r * sign(r)       (column sum)
//   g_rho[model] += g_y * (x - x_prev)   (scalar sum; clamp gate on host)
// grid (ceil(n/256), ceil(B/ROWS), M); block 256; coalesced row-major loops.
// ---------------------------------------------------------------------------
#define LBW_COLS 256
#define LBW_ROWS 256
extern "C" __global__ __launch_bounds__(LBW_COLS)
void k_lista_bwd_elem(const float* __restrict__ g_y,
                      const float* __restrict__ carry_in,  // nullptr on layer L
                      const float* __restrict__ r,
                      const float* __restrict__ theta,   // [M, n]
                      const float* __restrict__ x,
                      const float* __restrict__ x_prev,
                      const float* __restrict__ mom,     // [M]
                      float* __restrict__ g_r,
                      float* __restrict__ carry_out,
                      float* __restrict__ g_theta,       // [M, n] (pre-zeroed)
                      float* __restrict__ g_rho,         // [M]    (pre-zeroed)
                      int B, int n) {
  const int m = blockIdx.z;
  const int col = blockIdx.x * LBW_COLS + threadIdx.x;
  const int row0 = blockIdx.y * LBW_ROWS;
  const int row1 = min(row0 + LBW_ROWS, B);
  const bool col_ok = col < n;
  const float mm = mom[m];
  const float th = col_ok ? theta[(long)m * n + col] : 0.f;
  const long base = (long)m * B * n;

  float th_sum = 0.f, rho_sum = 0.f;
  if (col_ok) {
    for (int row = row0; row < row1; ++row) {
      long idx = base + (long)row * n + col;
      float gy = g_y[idx];
      float gx = (1.0f + mm) * gy + (carry_in ? carry_in[idx] : 0.f);
      float rv = r[idx];
      float gr = (fabsf(rv) > th) ? gx : 0.f;
      g_r[idx] = gr;
      carry_out[idx] = -mm * gy;
      th_sum -= gr * ((rv > 0.f) ? 1.f : ((rv < 0.f) ? -1.f : 0.f));
      rho_sum += gy * (x[idx] - x_prev[idx]);
    }
    if (th_sum != 0.f) atomicAdd(&g_theta[(long)m * n + col], th_sum);
  }
  rho_sum = wave_reduce_sum(rho_sum);
  __shared__ float warp_part[LBW_COLS / WAVE];
  if ((threadIdx.x & (WAVE - 1)) == 0) warp_part[threadIdx.x / WAVE] = rho_sum;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int w = 0; w < LBW_COLS / WAVE; ++w) s += warp_part[w];
    if (s != 0.f) atomicAdd(&g_rho[m], s);
  }
}

// ---------------------------------------------------------------------------
// k_topk_select: per-row top-k selection + scatter for the TopK encoder
// (SURVEY.md K8).  One 256-thread block per (model, batch-row); exact
// radix select over the order-preserving uint mapping of fp32 (4 passes of
// 8-bit histograms), then one write pass producing the dense code row
//   c[j] = j in top-k ? max(score[j], 0) : 0     (reference topk_encoder.py
// keeps clamped values in the selected slots), with fired counts fused.
// Exact-equal ties at the threshold are admitted in arbitrary order via an
// LDS slot counter (measure-zero for continuous scores).
// ---------------------------------------------------------------------------
#define TOPK_T 256
__device__ __forceinline__ unsigned f32_ord(float f) {
  unsigned u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

extern "C" __global__ __launch_bounds__(TOPK_T)
void k_topk_select(const float* __restrict__ scores,  // [M, B, n]
                   float* __restrict__ c_out,         // [M, B, n]
                   float* __restrict__ fired,         // [M, n]
                   const int* __restrict__ ks,        // [M]
                   int B, int n) {
  const int m = blockIdx.z;
  const int row = blockIdx.x;
  if (row >= B) return;
  const float* s = scores + ((long)m * B + row) * n;
  float* c = c_out + ((long)m * B + row) * n;
  float* fired_m = fired + (long)m * n;
  int k = ks[m];
  if (k >= n) k = n;

  __shared__ unsigned hist[256];
  __shared__ unsigned sh_prefix, sh_mask, sh_kleft, sh_slots;

  if (threadIdx.x == 0) {
    sh_prefix = 0u;
    sh_mask = 0u;  // bits of the prefix that are decided
    sh_kleft = (unsigned)k;
  }
  __syncthreads();

  // 4 radix passes, MSB first
  for (int shift = 24; shift >= 0; shift -= 8) {
    if (threadIdx.x < 256) hist[threadIdx.x] = 0u;
    __syncthreads();
    unsigned prefix = sh_prefix, mask = sh_mask;
    for (int j = threadIdx.x; j < n; j += TOPK_T) {
      unsigned u = f32_ord(s[j]);
      if ((u & mask) == prefix) atomicAdd(&hist[(u >> shift) & 0xFFu], 1u);
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      unsigned kleft = sh_kleft;
      int bin = 255;
      for (; bin >= 0; --bin) {
        if (hist[bin] >= kleft) break;
        kleft -= hist[bin];
      }
      if (bin < 0) bin = 0;  // defensive; cannot happen for k <= n
      sh_kleft = kleft;
      sh_prefix = sh_prefix | ((unsigned)bin << shift);
      sh_mask = sh_mask | (0xFFu << shift);
    }
    __syncthreads();
  }

  // threshold = exact k-th largest key; elements with key > thr are all
  // selected; sh_kleft of the == thr ties are selected in index order
  unsigned thr = sh_prefix;
  if (threadIdx.x == 0) sh_slots = sh_kleft;
  __syncthreads();
  for (int j = threadIdx.x; j < n; j += TOPK_T) {
    unsigned u = f32_ord(s[j]);
    float v = 0.f;
    bool sel = u > thr;
    if (!sel && u == thr) {
      unsigned slot = atomicSub(&sh_slots, 1u);
      sel = (slot != 0u && slot <= (unsigned)n);  // old value; wraps reject
    }
    if (sel) {
      v = fmaxf(s[j], 0.f);
      if (v > 0.f) atomicAdd(&fired_m[j], 1.f);
    }
    c[j] = v;
  }
}

// ---------------------------------------------------------------------------
// k_grad_w_t
// ---------------------------------------------------------------------------
template <int TBK, int MINW, int TBN = BN>
__global__ __launch_bounds__(NTHREADS, MINW)
void k_grad_w_t(const float* __restrict__ P, long p_batch_stride,
                const float* __restrict__ Q, long q_batch_stride,
                float* __restrict__ gw,  // [M, n, d]
                float alpha, float beta,
                int B, int n, int d, int prio) {
  __shared__ float As[2][TBK * BM];
  __shared__ float Bs[2][TBK * TBN];

  maybe_prio(prio);
  const int m = blockIdx.z;
  int tx, ty;
  tile_coords(tx, ty);
  const int row0 = ty * BM;  // n rows
  const int col0 = tx * TBN;  // d cols
  const float* P_m = P + (long)m * p_batch_stride;
  const float* Q_m = Q + (long)m * q_batch_stride;

  constexpr int NACC = TBN / 64;
  f32x16 acc[NACC];
  zero_acc<NACC>(acc);
  DStage<TBK> sa;
  DStage<TBK, TBN> sb;

#define GW_LA(K) stage_D_load<TBK>(P_m, n, (K), row0, B, n, nullptr, sa)
#define GW_LB(K) stage_D_load<TBK, TBN>(Q_m, d, (K), col0, B, d, nullptr, sb)
#define GW_WA(BUF) stage_D_write<TBK>(sa, &As[BUF][0], false)
#define GW_WB(BUF) stage_D_write<TBK, TBN>(sb, &Bs[BUF][0], false)
  PREFETCH_LOOP(TBK, B, GW_LA, GW_LB, GW_WA, GW_WB, BM, TBN)
#undef GW_LA
#undef GW_LB
#undef GW_WA
#undef GW_WB

  const EpiGeom g = epi_geom<NACC>();
  float* gw_m = gw + (long)m * n * d;

#pragma unroll
  for (int tj = 0; tj < NACC; ++tj) {
    int col = col0 + g.wc + tj * 32 + g.l31;
    bool col_ok = col < d;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int row = row0 + g.wr + acc_row(r, g.lane);
      if (row < n && col_ok) {
        long idx = (long)row * d + col;
        float v = alpha * acc[tj][r];
        if (beta != 0.f) v += beta * gw_m[idx];
        gw_m[idx] = v;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// k_project_adam: per dictionary row i of model m:
//   if project: g = (gw - (gw . w)/norm^2 * w * [norm>eps]) / max(norm,eps)
//   Adam with per-model bias correction.
// One wave per row; 8 rows per block.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(NTHREADS)
void k_project_adam(float* __restrict__ W,        // [M*n, d]
                    const float* __restrict__ gw, // [M*n, d]
                    const float* __restrict__ norms,  // [M*n]
                    float* __restrict__ mu, float* __restrict__ nu,
                    const float* __restrict__ step_no,  // [M]
                    int n_rows_total, int n_per_model, int d,
                    float lr, float b1, float b2, float eps_adam,
                    float eps_norm, int project,
                    const float* __restrict__ w_used_p,  // nullptr -> W
                    int clamp_mask,  // 1: g *= [W >= 0] (clamp backward)
                    const float* __restrict__ lr_mult) { // [M*n] per-row lr
                                                         // scale (post-resample
                                                         // warmup) or nullptr
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int row = blockIdx.x * (NTHREADS / WAVE) + wave;
  if (row >= n_rows_total) return;
  const float lr_eff = lr_mult ? lr * lr_mult[row] : lr;

  float* w_param = W + (long)row * d;
  const float* w = w_used_p ? w_used_p + (long)row * d : w_param;
  const float* g_in = gw + (long)row * d;
  float* mu_r = mu + (long)row * d;
  float* nu_r = nu + (long)row * d;

  const int m = row / n_per_model;
  const float t = step_no[m];
  const float bc1 = 1.0f - powf(b1, t);
  const float bc2 = 1.0f - powf(b2, t);

  float inv_s = 1.0f, dot_scaled = 0.f;
  bool do_proj = false;
  if (project) {
    float nrm = norms[row];
    float s = fmaxf(nrm, eps_norm);
    inv_s = 1.0f / s;
    do_proj = nrm > eps_norm;
    if (do_proj) {
      float acc = 0.f;
      for (int j = lane; j < d; j += WAVE) acc += g_in[j] * w[j];
      acc = wave_reduce_sum(acc);
      dot_scaled = acc / (nrm * nrm);
    }
  }

  for (int j = lane; j < d; j += WAVE) {
    float g = g_in[j];
    if (project) {
      if (do_proj) g = (g - dot_scaled * w[j]) * inv_s;
      else g = g * inv_s;
    }
    if (clamp_mask && w_param[j] < 0.f) g = 0.f;
    float m1 = b1 * mu_r[j] + (1.0f - b1) * g;
    float v1 = b2 * nu_r[j] + (1.0f - b2) * g * g;
    mu_r[j] = m1;
    nu_r[j] = v1;
    w_param[j] -= lr_eff * (m1 / bc1) / (sqrtf(v1 / bc2) + eps_adam);
  }
}

// ---------------------------------------------------------------------------
// k_bias_adam: Adam on [M, n] bias with optional L2-norm decay gradient.
// grid (M); block NTHREADS.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(NTHREADS)
void k_bias_adam(float* __restrict__ bias,        // [M, n]
                 const float* __restrict__ g_bias,// [M, n]
                 const float* __restrict__ bias_decay, // [M]
                 float* __restrict__ mu, float* __restrict__ nu,
                 const float* __restrict__ step_no,
                 int n, float lr, float b1, float b2, float eps_adam,
                 const float* __restrict__ lr_mult) { // [M, n] or nullptr
  const int m = blockIdx.x;
  const float* lrm_m = lr_mult ? lr_mult + (long)m * n : nullptr;
  float* b_m = bias + (long)m * n;
  const float* g_m = g_bias + (long)m * n;
  float* mu_m = mu + (long)m * n;
  float* nu_m = nu + (long)m * n;
  const float bd = bias_decay[m];
  const float t = step_no[m];
  const float bc1 = 1.0f - powf(b1, t);
  const float bc2 = 1.0f - powf(b2, t);

  __shared__ float partial[NTHREADS / WAVE];
  __shared__ float norm_sq_s;
  float decay_scale = 0.f;
  if (bd != 0.f) {
    float acc = 0.f;
    for (int j = threadIdx.x; j < n; j += NTHREADS) acc += b_m[j] * b_m[j];
    acc = wave_reduce_sum(acc);
    if ((threadIdx.x & (WAVE - 1)) == 0) partial[threadIdx.x / WAVE] = acc;
    __syncthreads();
    if (threadIdx.x == 0) {
      float s = 0.f;
      for (int wv = 0; wv < NTHREADS / WAVE; ++wv) s += partial[wv];
      norm_sq_s = s;
    }
    __syncthreads();
    float nrm = sqrtf(norm_sq_s);
    decay_scale = (nrm > 0.f) ? bd / nrm : 0.f;
  }

  for (int j = threadIdx.x; j < n; j += NTHREADS) {
    float g = g_m[j] + decay_scale * b_m[j];
    float m1 = b1 * mu_m[j] + (1.0f - b1) * g;
    float v1 = b2 * nu_m[j] + (1.0f - b2) * g * g;
    mu_m[j] = m1;
    nu_m[j] = v1;
    float lr_eff = lrm_m ? lr * lrm_m[j] : lr;
    b_m[j] -= lr_eff * (m1 / bc1) / (sqrtf(v1 / bc2) + eps_adam);
  }
}

// ---------------------------------------------------------------------------
// k_resample: the dead-neuron resampling rule fully fused:
// one block per model scans the fired counters, ranks dead features in
// index order (block-wide ballot prefix over 256-column segments), and for
// the first n_track of them overwrites the encoder row with the matching
// worst-example direction (scaled), zeroes the Adam moments, resets the
// bias (+its moments), and optionally rewrites a decoder row.  No host loop,
// no per-model syncs; counts land in counts_out.
// ---------------------------------------------------------------------------
#define RSMP_T 256
extern "C" __global__ __launch_bounds__(RSMP_T)
void k_resample(const float* __restrict__ fired,      // [M, n]
                const float* __restrict__ new_rows,   // [M, T, d] unit dirs
                const float* __restrict__ enc_scale,  // [M] row-norm scale
                float* __restrict__ W,                // [M, n, d] encoder
                float* __restrict__ mu_w, float* __restrict__ nu_w,
                float* __restrict__ dec,              // [M, n, d] or nullptr
                float* __restrict__ mu_d, float* __restrict__ nu_d,
                float* __restrict__ bias,             // [M, n] or nullptr
                float* __restrict__ mu_b, float* __restrict__ nu_b,
                int* __restrict__ counts_out,         // [M]
                int n, int d, int n_track) {
  const int m = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wave = tid / WAVE;
  const float* fired_m = fired + (long)m * n;
  const float scale = enc_scale[m];

  __shared__ int base;
  __shared__ int wave_cnt[RSMP_T / WAVE];
  if (tid == 0) base = 0;
  __syncthreads();

  for (int s0 = 0; s0 < n; s0 += RSMP_T) {
    int j = s0 + tid;
    bool isdead = (j < n) && (fired_m[j] == 0.0f);
    unsigned long long mask = __ballot(isdead);
    int lane_rank = __popcll(mask & ((1ull << lane) - 1ull));
    if (lane == 0) wave_cnt[wave] = __popcll(mask);
    __syncthreads();
    int wave_prefix = 0;
    for (int w = 0; w < wave; ++w) wave_prefix += wave_cnt[w];
    int rank = base + wave_prefix + lane_rank;
    if (isdead && rank < n_track) {
      const float* src = new_rows + ((long)m * n_track + rank) * d;
      float* w_row = W + ((long)m * n + j) * d;
      float* muw = mu_w + ((long)m * n + j) * d;
      float* nuw = nu_w + ((long)m * n + j) * d;
      for (int t = 0; t < d; ++t) {
        float v = src[t];
        w_row[t] = v * scale;
        muw[t] = 0.f;
        nuw[t] = 0.f;
      }
      if (dec) {
        float* d_row = dec + ((long)m * n + j) * d;
        float* mud = mu_d + ((long)m * n + j) * d;
        float* nud = nu_d + ((long)m * n + j) * d;
        for (int t = 0; t < d; ++t) {
          d_row[t] = src[t];
          mud[t] = 0.f;
          nud[t] = 0.f;
        }
      }
      if (bias) {
        bias[(long)m * n + j] = 0.f;
        mu_b[(long)m * n + j] = 0.f;
        nu_b[(long)m * n + j] = 0.f;
      }
    }
    __syncthreads();
    if (tid == 0) {
      int tot = 0;
      for (int w = 0; w < RSMP_T / WAVE; ++w) tot += wave_cnt[w];
      base += tot;
    }
    __syncthreads();
    if (base >= n_track && s0 + RSMP_T < n) {
      // everything past here would exceed the replacement budget; still
      // need the total dead count? reference reports replaced count only.
      break;
    }
  }
  if (tid == 0) counts_out[m] = base < n_track ? base : n_track;
}

// ---------------------------------------------------------------------------
// k_colsum: out[m, j] = alpha * sum_b X[m, b, j] for X [M, B, n].
// The column-sum tail of the residual/semilinear backward: torch's strided
// reduce runs this at ~140 GB/s (profiles/r02_*_kernel_stats.csv); here each
// 256-thread block owns 256 consecutive columns of one (model, B-slice) so
// every row read is one fully-coalesced 1 KB line burst, and the B axis is
// split over grid.z (partials combined with atomicAdd; caller zero-fills
// out) to keep >= 4k waves in flight on the 256-CU chip.
#define COLSUM_T 256
extern "C" __global__ __launch_bounds__(COLSUM_T)
void k_colsum(const float* __restrict__ X, float* __restrict__ out,
              int B, int n, float alpha, int absval) {
  const int j = blockIdx.x * COLSUM_T + threadIdx.x;
  const int m = blockIdx.y;
  const int nsplit = gridDim.z;
  if (j >= n) return;
  const long base = (long)m * B * n;
  const int b0 = (int)(((long)blockIdx.z * B) / nsplit);
  const int b1 = (int)(((long)(blockIdx.z + 1) * B) / nsplit);
  float acc = 0.f;
  if (absval) {
    #pragma unroll 4
    for (int b = b0; b < b1; ++b) acc += fabsf(X[base + (long)b * n + j]);
  } else {
    #pragma unroll 4
    for (int b = b0; b < b1; ++b) acc += X[base + (long)b * n + j];
  }
  atomicAdd(&out[(long)m * n + j], acc * alpha);
}

// ---------------------------------------------------------------------------
// k_transpose_scale: dst[c][r] = src[r][c] * (scale ? scale[r] : 1), batched
// over grid.z with explicit strides.  64x64 LDS tiles, coalesced both sides.
// Feeds the all-direct-staged GEMM variants below (x^T, r^T, What^T).
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256)
void k_transpose_scale(const float* __restrict__ src, float* __restrict__ dst,
                       const float* __restrict__ scale,
                       int R, int C,
                       long src_mstride, long dst_mstride, long scale_mstride) {
  __shared__ float tile[64][65];
  const int m = blockIdx.z;
  src += (long)m * src_mstride;
  dst += (long)m * dst_mstride;
  if (scale) scale += (long)m * scale_mstride;
  const int r0 = blockIdx.y * 64;
  const int c0 = blockIdx.x * 64;
  const int t = threadIdx.x;

#pragma unroll
  for (int p = 0; p < 16; ++p) {
    int r = p * 4 + t / 64;
    int c = t % 64;
    float v = 0.f;
    if (r0 + r < R && c0 + c < C) {
      v = src[(long)(r0 + r) * C + c0 + c];
      if (scale) v *= scale[r0 + r];
    }
    tile[r][c] = v;
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 16; ++p) {
    int c = p * 4 + t / 64;  // output row (= source column)
    int r = t % 64;          // output col (= source row)
    if (c0 + c < C && r0 + r < R) dst[(long)(c0 + c) * R + r0 + r] = tile[r][c];
  }
}

// ---------------------------------------------------------------------------
// k_enc_fwd2_t: enc forward with PRE-TRANSPOSED operands (xT [d,B], WT [M,d,n],
// already inv-norm-scaled for tied) — both tiles stage DIRECT (b128 writes).
// ---------------------------------------------------------------------------
template <int TBK, int MINW, int TBN = BN>
__global__ __launch_bounds__(NTHREADS, MINW)
void k_enc_fwd2_t(const float* __restrict__ xT,      // [d, B]
                  const float* __restrict__ WT,      // [M, d, n]
                  const float* __restrict__ bias,    // [M, n]
                  float* __restrict__ c_out,         // [M, B, n]
                  float* __restrict__ loss_parts,    // [M, 2]
                  float* __restrict__ fired,         // [M, n]
                  int B, int d, int n, int mode, int prio,
                  const int* __restrict__ dict_sizes) {
  __shared__ float As[2][TBK * BM];
  __shared__ float Bs[2][TBK * TBN];

  maybe_prio(prio);
  const int m = blockIdx.z;
  int tx, ty;
  tile_coords(tx, ty);
  const int row0 = ty * BM;   // batch rows
  const int col0 = tx * TBN;   // dict cols
  const float* WT_m = WT + (long)m * d * n;

  constexpr int NACC = TBN / 64;
  f32x16 acc[NACC];
  zero_acc<NACC>(acc);
  DStage<TBK> sa;
  DStage<TBK, TBN> sb;

#define ENC2_LA(K) stage_D_load<TBK>(xT, B, (K), row0, d, B, nullptr, sa)
#define ENC2_LB(K) stage_D_load<TBK, TBN>(WT_m, n, (K), col0, d, n, nullptr, sb)
#define ENC2_WA(BUF) stage_D_write<TBK>(sa, &As[BUF][0], false)
#define ENC2_WB(BUF) stage_D_write<TBK, TBN>(sb, &Bs[BUF][0], false)
  PREFETCH_LOOP(TBK, d, ENC2_LA, ENC2_LB, ENC2_WA, ENC2_WB, BM, TBN)
#undef ENC2_LA
#undef ENC2_LB
#undef ENC2_WA
#undef ENC2_WB

  const EpiGeom g = epi_geom<NACC>();
  float* c_m = c_out + (long)m * B * n;
  const float* bias_m = bias + (long)m * n;
  float* fired_m = fired + (long)m * n;

  float l1_sum = 0.f;
#pragma unroll
  for (int tj = 0; tj < NACC; ++tj) {
    int col = col0 + g.wc + tj * 32 + g.l31;
    bool col_ok = col < n;
    if (mode == 1) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int row = row0 + g.wr + acc_row(r, g.lane);
        if (row < B && col_ok) c_m[(long)row * n + col] = acc[tj][r];
      }
      continue;
    }
    float bj = col_ok ? bias_m[col] : 0.f;
    bool live = col_ok && (!dict_sizes || col < dict_sizes[m]);
    float fired_cnt = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int row = row0 + g.wr + acc_row(r, g.lane);
      if (row < B && col_ok) {
        float v = live ? fmaxf(acc[tj][r] + bj, 0.f) : 0.f;
        c_m[(long)row * n + col] = v;
        l1_sum += v;
        fired_cnt += (v > 0.f) ? 1.f : 0.f;
      }
    }
    if (col_ok) {
      float other = __shfl_xor(fired_cnt, 32, WAVE);
      float tot = fired_cnt + other;
      if (g.lane < 32 && tot > 0.f) atomicAdd(&fired_m[col], tot);
    }
  }
  if (mode == 0) {
    l1_sum = wave_reduce_sum(l1_sum);
    if (g.lane == 0) atomicAdd(&loss_parts[m * 2 + 1], l1_sum);
  }
}

// ---------------------------------------------------------------------------
// k_gc2_t: code-grad with pre-transposed rT [M,d,B] and WT (scaled) [M,d,n].
// ---------------------------------------------------------------------------
template <int TBK, int MINW, int TBN = BN>
__global__ __launch_bounds__(NTHREADS, MINW)
void k_gc2_t(const float* __restrict__ rT,       // [M, d, B]
             const float* __restrict__ WT,       // [M, d, n] (inv-norm scaled)
             const float* __restrict__ c,        // [M, B, n]
             const float* __restrict__ l1_alpha, // [M]
             float* __restrict__ gpre_out,       // [M, B, n]
             float* __restrict__ g_bias,         // [M, n]
             int B, int d, int n, int prio, int gc_mode) {
  __shared__ float As[2][TBK * BM];
  __shared__ float Bs[2][TBK * TBN];

  maybe_prio(prio);
  const int m = blockIdx.z;
  int tx, ty;
  tile_coords(tx, ty);
  const int row0 = ty * BM;
  const int col0 = tx * TBN;
  const float* rT_m = rT + (long)m * d * B;
  const float* WT_m = WT + (long)m * d * n;
  const float* c_m = c + (long)m * B * n;
  const float gscale = 2.0f / ((float)B * (float)d);
  const float l1_term = l1_alpha[m] / (float)B;

  constexpr int NACC = TBN / 64;
  f32x16 acc[NACC];
  zero_acc<NACC>(acc);
  DStage<TBK> sa;
  DStage<TBK, TBN> sb;

#define GC2_LA(K) stage_D_load<TBK>(rT_m, B, (K), row0, d, B, nullptr, sa)
#define GC2_LB(K) stage_D_load<TBK, TBN>(WT_m, n, (K), col0, d, n, nullptr, sb)
#define GC2_WA(BUF) stage_D_write<TBK>(sa, &As[BUF][0], false)
#define GC2_WB(BUF) stage_D_write<TBK, TBN>(sb, &Bs[BUF][0], false)
  PREFETCH_LOOP(TBK, d, GC2_LA, GC2_LB, GC2_WA, GC2_WB, BM, TBN)
#undef GC2_LA
#undef GC2_LB
#undef GC2_WA
#undef GC2_WB

  const EpiGeom g = epi_geom<NACC>();
  float* g_m = gpre_out + (long)m * B * n;
  float* gb_m = g_bias + (long)m * n;

#pragma unroll
  for (int tj = 0; tj < NACC; ++tj) {
    int col = col0 + g.wc + tj * 32 + g.l31;
    bool col_ok = col < n;
    float colsum = 0.f;
#pragma unroll
    for (int r_ = 0; r_ < 16; ++r_) {
      int row = row0 + g.wr + acc_row(r_, g.lane);
      if (row < B && col_ok) {
        float cv = c_m[(long)row * n + col];
        float gv;
        if (gc_mode == 1) {
          float sgn = (cv > 0.f) ? 1.f : ((cv < 0.f) ? -1.f : 0.f);
          gv = (cv != 0.f) ? (gscale * acc[tj][r_] + l1_term * sgn) : 0.f;
        } else {
          gv = (cv > 0.f) ? (gscale * acc[tj][r_] + l1_term) : 0.f;
        }
        g_m[(long)row * n + col] = gv;
        colsum += gv;
      }
    }
    if (col_ok && gc_mode == 0) {
      float other = __shfl_xor(colsum, 32, WAVE);
      float tot = colsum + other;
      if (g.lane < 32 && tot != 0.f) atomicAdd(&gb_m[col], tot);
    }
  }
}
