// CDNA4 (gfx950) kernels for SAE-ensemble training, fp32 end to end.
//
// These implement the reference's implicit kernel inventory (SURVEY.md §2.4
// K1-K6: encoder GEMM -> ReLU -> decoder GEMM -> MSE+L1 forward, analytic
// backward through the in-forward decoder row renormalization, fused Adam)
// as hand-written MFMA kernels.  fp32 is the reference's training dtype
// (BASELINE.md); we use the exact-f32 matrix instruction
// v_mfma_f32_32x32x2_f32 (157 TF peak, bitwise == fmaf chain).
//
// Structure: one block-tile GEMM core (128x128 output tile, BK=32, 4 waves,
// each wave a 64x64 quadrant as 2x2 v_mfma_f32_32x32x2_f32 accumulators),
// instantiated with different operand-staging modes and fused epilogues:
//   k_enc_fwd : c = relu(x @ Wenc^T + b)    (+ L1 partial, fired counts)
//   k_dec_fwd : r = c @ Wdec_hat - x        (+ MSE partial)
//   k_gc      : gpre = (c>0) .* (gs * r @ Wdec_hat^T + l1/B)  (+ bias-grad)
//   k_grad_w  : gw = beta*gw + alpha * P^T @ Q   (K = batch contraction)
// plus k_row_norms (dictionary row norms), k_project_adam (gradient of
// w/max(||w||,eps) + Adam), k_bias_adam.
//
// LDS: transposed-stage tiles use stride BM+1 (conflict-free b32 writes and
// reads); direct-stage tiles use stride BM with float4 (ds_write_b128)
// writes (8-lane groups cover all 32 banks).  Grids are (tiles_n, tiles_m,
// n_models) -- thousands of workgroups for the flagship shapes, enough to
// fill 256 CUs across 8 XCDs.

#include <hip/hip_runtime.h>

#define WAVE 64
#define BM 128
#define BN 128
#define BK 32
#define BMP (BM + 1)  // padded LDS stride for transposed staging
#define NTHREADS 256

typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef float f32x4n __attribute__((ext_vector_type(4)));

// C/D fragment mapping for 32x32 MFMA (guide §3): reg r, lane l ->
//   row = (r&3) + 8*(r>>2) + 4*(l>>5), col = l&31
__device__ __forceinline__ int acc_row(int r, int lane) {
  return (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
}

// ---------------------------------------------------------------------------
// staging
// ---------------------------------------------------------------------------

// Transposed stage: lds[k][i] = src[i0+i][k0+k] for i in [0,ROWS), k in [0,BK)
// src is row-major [n_rows, ld].  Guarded (zero-fill outside).  Conflict-free
// by the +1 pad: bank(lds[k][i]) = (k*BMP + i) % 32 = (k + i) % 32 and each
// 32-lane group covers (4*kidx + s + i) = all residues once.
template <int ROWS>
__device__ __forceinline__ void stage_T(const float* __restrict__ src, long ld,
                                        int i0, int k0, int n_rows, int n_k,
                                        float* __restrict__ lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int p = 0; p < ROWS * BK / (NTHREADS * 4); ++p) {
    int i = p * (NTHREADS / 8) + t / 8;   // 32 rows per pass
    int kc = (t % 8) * 4;
    int gi = i0 + i;
    float4 v = make_float4(0.f, 0.f, 0.f, 0.f);
    if (gi < n_rows) {
      const float* row = src + (long)gi * ld + k0 + kc;
      // guard k range
      if (k0 + kc + 3 < n_k) {
        v = *reinterpret_cast<const float4*>(row);
      } else {
#pragma unroll
        for (int s = 0; s < 4; ++s)
          if (k0 + kc + s < n_k) ((float*)&v)[s] = row[s];
      }
    }
#pragma unroll
    for (int s = 0; s < 4; ++s) lds[(kc + s) * BMP + i] = ((float*)&v)[s];
  }
}

// Direct stage: lds[k][j] = src[k0+k][j0+j] * (scale ? scale[k0+k] : 1)
// stride BM, float4 writes (16B-aligned, conflict-free for b128).
template <int COLS>
__device__ __forceinline__ void stage_D(const float* __restrict__ src, long ld,
                                        int k0, int j0, int n_k, int n_cols,
                                        const float* __restrict__ scale,
                                        float* __restrict__ lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int p = 0; p < BK * COLS / (NTHREADS * 4); ++p) {
    int k = p * (NTHREADS / 32) + t / 32;  // 8 k-rows per pass
    int j = (t % 32) * 4;
    int gk = k0 + k;
    float4 v = make_float4(0.f, 0.f, 0.f, 0.f);
    if (gk < n_k) {
      const float* row = src + (long)gk * ld + j0 + j;
      if (j0 + j + 3 < n_cols) {
        v = *reinterpret_cast<const float4*>(row);
      } else {
#pragma unroll
        for (int s = 0; s < 4; ++s)
          if (j0 + j + s < n_cols) ((float*)&v)[s] = row[s];
      }
      if (scale) {
        float sc = scale[gk];
        v.x *= sc; v.y *= sc; v.z *= sc; v.w *= sc;
      }
    }
    *reinterpret_cast<float4*>(&lds[k * BM + j]) = v;
  }
}

// Apply a per-row (j-indexed) scale to a transposed-staged tile's source:
// done at read time would cost per-MFMA VALU; instead scale during staging.
template <int ROWS>
__device__ __forceinline__ void stage_T_scaled(const float* __restrict__ src, long ld,
                                               int i0, int k0, int n_rows, int n_k,
                                               const float* __restrict__ scale,
                                               float* __restrict__ lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int p = 0; p < ROWS * BK / (NTHREADS * 4); ++p) {
    int i = p * (NTHREADS / 8) + t / 8;
    int kc = (t % 8) * 4;
    int gi = i0 + i;
    float4 v = make_float4(0.f, 0.f, 0.f, 0.f);
    if (gi < n_rows) {
      const float* row = src + (long)gi * ld + k0 + kc;
      if (k0 + kc + 3 < n_k) {
        v = *reinterpret_cast<const float4*>(row);
      } else {
#pragma unroll
        for (int s = 0; s < 4; ++s)
          if (k0 + kc + s < n_k) ((float*)&v)[s] = row[s];
      }
      float sc = scale[gi];
      v.x *= sc; v.y *= sc; v.z *= sc; v.w *= sc;
    }
#pragma unroll
    for (int s = 0; s < 4; ++s) lds[(kc + s) * BMP + i] = ((float*)&v)[s];
  }
}

// ---------------------------------------------------------------------------
// the MFMA block loop
// ---------------------------------------------------------------------------
// As: [BK][BMP or BM], Bs: [BK][BMP or BM].  Each of 4 waves computes the
// 64x64 quadrant (wr, wc); acc[ti][tj] is the (32x32) sub-tile.
template <int ASTRIDE, int BSTRIDE>
__device__ __forceinline__ void mfma_tile(const float* __restrict__ As,
                                          const float* __restrict__ Bs,
                                          f32x16 acc[2][2]) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;
  const int l31 = lane & 31;
  const int h = lane >> 5;  // k sub-index

#pragma unroll
  for (int kk = 0; kk < BK; kk += 2) {
    float a0 = As[(kk + h) * ASTRIDE + wr + l31];
    float a1 = As[(kk + h) * ASTRIDE + wr + 32 + l31];
    float b0 = Bs[(kk + h) * BSTRIDE + wc + l31];
    float b1 = Bs[(kk + h) * BSTRIDE + wc + 32 + l31];
    acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
    acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
    acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
    acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
  }
}

__device__ __forceinline__ void zero_acc(f32x16 acc[2][2]) {
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[i][j][r] = 0.f;
}

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

// ---------------------------------------------------------------------------
// k_row_norms: norms[m][i] = ||W[m][i][:]||_2 ; inv[m][i] = 1/max(norm, eps)
// ---------------------------------------------------------------------------
extern "C" __global__ void k_row_norms(const float* __restrict__ W,
                                       float* __restrict__ norms,
                                       float* __restrict__ inv_norms,
                                       int n_rows_total, int d, float eps) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int row = blockIdx.x * 4 + wave;
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
// k_enc_fwd: out[b, j] = relu( sum_k x[b,k] * Wenc[j,k] * (tied? inv[j]:1) + bias[j] )
// grid: (ceil(n/BN), ceil(B/BM), M)
// epilogue extras: l1 partial sum -> loss_parts[m*2+1]; fired[m][j] += count
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(NTHREADS)
void k_enc_fwd(const float* __restrict__ x,       // [B, d]
               const float* __restrict__ Wenc,    // [M, n, d]
               const float* __restrict__ bias,    // [M, n]
               const float* __restrict__ inv_norms, // [M, n] or nullptr
               float* __restrict__ c_out,         // [M, B, n]
               float* __restrict__ loss_parts,    // [M, 2]
               float* __restrict__ fired,         // [M, n]
               int B, int d, int n) {
  __shared__ float As[BK * BMP];
  __shared__ float Bs[BK * BMP];

  const int m = blockIdx.z;
  const int row0 = blockIdx.y * BM;   // batch rows
  const int col0 = blockIdx.x * BN;   // dict rows (output cols)
  const float* W = Wenc + (long)m * n * d;
  const float* inv = inv_norms ? inv_norms + (long)m * n : nullptr;

  f32x16 acc[2][2];
  zero_acc(acc);

  for (int k0 = 0; k0 < d; k0 += BK) {
    stage_T<BM>(x, d, row0, k0, B, d, As);
    if (inv)
      stage_T_scaled<BN>(W, d, col0, k0, n, d, inv, Bs);
    else
      stage_T<BN>(W, d, col0, k0, n, d, Bs);
    __syncthreads();
    mfma_tile<BMP, BMP>(As, Bs, acc);
    __syncthreads();
  }

  // epilogue: bias add, relu, store, l1 partial, fired counts
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;
  const int l31 = lane & 31;
  float* c_m = c_out + (long)m * B * n;
  const float* bias_m = bias + (long)m * n;
  float* fired_m = fired + (long)m * n;

  float l1_sum = 0.f;
#pragma unroll
  for (int tj = 0; tj < 2; ++tj) {
    int col = col0 + wc + tj * 32 + l31;
    bool col_ok = col < n;
    float bj = col_ok ? bias_m[col] : 0.f;
    float fired_cnt = 0.f;
#pragma unroll
    for (int ti = 0; ti < 2; ++ti) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int row = row0 + wr + ti * 32 + acc_row(r, lane);
        if (row < B && col_ok) {
          float v = acc[ti][tj][r] + bj;
          v = fmaxf(v, 0.f);
          c_m[(long)row * n + col] = v;
          l1_sum += v;
          fired_cnt += (v > 0.f) ? 1.f : 0.f;
        }
      }
    }
    if (col_ok && fired_cnt > 0.f) {
      // combine the two half-wave contributions for this column? lanes l and
      // l+32 hold DIFFERENT rows of the SAME column: merge via shfl to halve
      // atomics
      float other = __shfl_xor(fired_cnt, 32, WAVE);
      if (lane < 32) atomicAdd(&fired_m[col], fired_cnt + other);
    }
  }
  l1_sum = wave_reduce_sum(l1_sum);
  if (lane == 0) atomicAdd(&loss_parts[m * 2 + 1], l1_sum);
}

// ---------------------------------------------------------------------------
// k_dec_fwd: r[b, j] = sum_k c[b,k] * Wdec[k,j] * inv[k]  -  x[b,j]
// contraction over n (k index), output [B, d].
// grid: (ceil(d/BN), ceil(B/BM), M).  epilogue: r store + MSE partial.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(NTHREADS)
void k_dec_fwd(const float* __restrict__ c,       // [M, B, n]
               const float* __restrict__ Wdec,    // [M, n, d]
               const float* __restrict__ inv_norms, // [M, n]
               const float* __restrict__ x,       // [B, d]
               float* __restrict__ r_out,         // [M, B, d]
               float* __restrict__ loss_parts,    // [M, 2]
               int B, int d, int n) {
  __shared__ float As[BK * BMP];
  __shared__ float Bs[BK * BM];

  const int m = blockIdx.z;
  const int row0 = blockIdx.y * BM;  // batch rows
  const int col0 = blockIdx.x * BN;  // d cols
  const float* c_m = c + (long)m * B * n;
  const float* W = Wdec + (long)m * n * d;
  const float* inv = inv_norms + (long)m * n;

  f32x16 acc[2][2];
  zero_acc(acc);

  for (int k0 = 0; k0 < n; k0 += BK) {
    stage_T<BM>(c_m, n, row0, k0, B, n, As);
    stage_D<BN>(W, d, k0, col0, n, d, inv, Bs);
    __syncthreads();
    mfma_tile<BMP, BM>(As, Bs, acc);
    __syncthreads();
  }

  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;
  const int l31 = lane & 31;
  float* r_m = r_out + (long)m * B * d;

  float mse_sum = 0.f;
#pragma unroll
  for (int tj = 0; tj < 2; ++tj) {
    int col = col0 + wc + tj * 32 + l31;
    bool col_ok = col < d;
#pragma unroll
    for (int ti = 0; ti < 2; ++ti) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int row = row0 + wr + ti * 32 + acc_row(r, lane);
        if (row < B && col_ok) {
          float rv = acc[ti][tj][r] - x[(long)row * d + col];
          r_m[(long)row * d + col] = rv;
          mse_sum += rv * rv;
        }
      }
    }
  }
  mse_sum = wave_reduce_sum(mse_sum);
  if (lane == 0) atomicAdd(&loss_parts[m * 2 + 0], mse_sum);
}

// ---------------------------------------------------------------------------
// k_gc: gpre[b, j] = (c[b,j] > 0) * ( gscale * sum_k r[b,k]*Wdec[j,k]*inv[j]
//                                     + l1_alpha[m] / B )
// grid: (ceil(n/BN), ceil(B/BM), M).  epilogue also accumulates
// g_bias[m][j] += sum_b gpre[b, j] over this row-tile.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(NTHREADS)
void k_gc(const float* __restrict__ r,        // [M, B, d]
          const float* __restrict__ Wdec,     // [M, n, d]
          const float* __restrict__ inv_norms,// [M, n]
          const float* __restrict__ c,        // [M, B, n]
          const float* __restrict__ l1_alpha, // [M]
          float* __restrict__ gpre_out,       // [M, B, n]
          float* __restrict__ g_bias,         // [M, n]
          int B, int d, int n) {
  __shared__ float As[BK * BMP];
  __shared__ float Bs[BK * BMP];

  const int m = blockIdx.z;
  const int row0 = blockIdx.y * BM;
  const int col0 = blockIdx.x * BN;
  const float* r_m = r + (long)m * B * d;
  const float* W = Wdec + (long)m * n * d;
  const float* inv = inv_norms + (long)m * n;
  const float* c_m = c + (long)m * B * n;
  const float gscale = 2.0f / ((float)B * (float)d);
  const float l1_term = l1_alpha[m] / (float)B;

  f32x16 acc[2][2];
  zero_acc(acc);

  for (int k0 = 0; k0 < d; k0 += BK) {
    stage_T<BM>(r_m, d, row0, k0, B, d, As);
    stage_T_scaled<BN>(W, d, col0, k0, n, d, inv, Bs);
    __syncthreads();
    mfma_tile<BMP, BMP>(As, Bs, acc);
    __syncthreads();
  }

  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;
  const int l31 = lane & 31;
  float* g_m = gpre_out + (long)m * B * n;
  float* gb_m = g_bias + (long)m * n;

#pragma unroll
  for (int tj = 0; tj < 2; ++tj) {
    int col = col0 + wc + tj * 32 + l31;
    bool col_ok = col < n;
    float colsum = 0.f;
#pragma unroll
    for (int ti = 0; ti < 2; ++ti) {
#pragma unroll
      for (int r_ = 0; r_ < 16; ++r_) {
        int row = row0 + wr + ti * 32 + acc_row(r_, lane);
        if (row < B && col_ok) {
          float cv = c_m[(long)row * n + col];
          float g = (cv > 0.f) ? (gscale * acc[ti][tj][r_] + l1_term) : 0.f;
          g_m[(long)row * n + col] = g;
          colsum += g;
        }
      }
    }
    if (col_ok) {
      float other = __shfl_xor(colsum, 32, WAVE);
      if (lane < 32) {
        float tot = colsum + other;
        if (tot != 0.f) atomicAdd(&gb_m[col], tot);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// k_grad_w: gw[i, j] = beta * gw[i, j] + alpha * sum_b P[b, i] * Q[b, j]
// P: [M, B, n] (i over n), Q: [M, B, d] or shared [B, d] (j over d).
// Both operands stage DIRECT (contraction index b is the row index of both).
// grid: (ceil(d/BN), ceil(n/BM), M)
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(NTHREADS)
void k_grad_w(const float* __restrict__ P, long p_batch_stride,
              const float* __restrict__ Q, long q_batch_stride,
              float* __restrict__ gw,  // [M, n, d]
              float alpha, float beta,
              int B, int n, int d) {
  __shared__ float As[BK * BM];
  __shared__ float Bs[BK * BM];

  const int m = blockIdx.z;
  const int row0 = blockIdx.y * BM;  // n rows of gw
  const int col0 = blockIdx.x * BN;  // d cols
  const float* P_m = P + (long)m * p_batch_stride;
  const float* Q_m = Q + (long)m * q_batch_stride;

  f32x16 acc[2][2];
  zero_acc(acc);

  for (int k0 = 0; k0 < B; k0 += BK) {
    // As[k][i] = P[k0+k][row0+i]
    stage_D<BM>(P_m, n, k0, row0, B, n, nullptr, As);
    // Bs[k][j] = Q[k0+k][col0+j]
    stage_D<BN>(Q_m, d, k0, col0, B, d, nullptr, Bs);
    __syncthreads();
    mfma_tile<BM, BM>(As, Bs, acc);
    __syncthreads();
  }

  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;
  const int l31 = lane & 31;
  float* gw_m = gw + (long)m * n * d;

#pragma unroll
  for (int tj = 0; tj < 2; ++tj) {
    int col = col0 + wc + tj * 32 + l31;
    bool col_ok = col < d;
#pragma unroll
    for (int ti = 0; ti < 2; ++ti) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int row = row0 + wr + ti * 32 + acc_row(r, lane);
        if (row < n && col_ok) {
          long idx = (long)row * d + col;
          float v = alpha * acc[ti][tj][r];
          if (beta != 0.f) v += beta * gw_m[idx];
          gw_m[idx] = v;
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// k_project_adam: per dictionary row i of model m:
//   if project: g = (gw - (gw . w_hat) * w_hat * [norm>eps]) / max(norm,eps)
//               (exact gradient of w / max(||w||, eps))
//   Adam: mu = b1 mu + (1-b1) g ; nu = b2 nu + (1-b2) g^2
//         w -= lr * (mu/(1-b1^t)) / (sqrt(nu/(1-b2^t)) + eps_adam)
// One wave per row; 4 rows per block.  grid: (ceil(M*n/4))
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(NTHREADS)
void k_project_adam(float* __restrict__ W,        // [M*n, d]
                    const float* __restrict__ gw, // [M*n, d]
                    const float* __restrict__ norms,  // [M*n]
                    float* __restrict__ mu, float* __restrict__ nu,
                    const float* __restrict__ step_no,  // [M] (post-increment)
                    int n_rows_total, int n_per_model, int d,
                    float lr, float b1, float b2, float eps_adam,
                    float eps_norm, int project) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int row = blockIdx.x * 4 + wave;
  if (row >= n_rows_total) return;

  float* w = W + (long)row * d;
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
      // dot = (g . w) / norm^2  (so that g_proj = (g - dot * w) / s)
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
    float m1 = b1 * mu_r[j] + (1.0f - b1) * g;
    float v1 = b2 * nu_r[j] + (1.0f - b2) * g * g;
    mu_r[j] = m1;
    nu_r[j] = v1;
    float upd = lr * (m1 / bc1) / (sqrtf(v1 / bc2) + eps_adam);
    w[j] -= upd;
  }
}

// ---------------------------------------------------------------------------
// k_bias_adam: Adam on the [M, n] bias with optional L2-norm decay gradient
//   g = g_bias + bias_decay[m] * b / ||b||   (term skipped when ||b|| == 0)
// grid: (M); block 256; each block handles one model's bias vector.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(NTHREADS)
void k_bias_adam(float* __restrict__ bias,        // [M, n]
                 const float* __restrict__ g_bias,// [M, n]
                 const float* __restrict__ bias_decay, // [M]
                 float* __restrict__ mu, float* __restrict__ nu,
                 const float* __restrict__ step_no,
                 int n, float lr, float b1, float b2, float eps_adam) {
  const int m = blockIdx.x;
  float* b_m = bias + (long)m * n;
  const float* g_m = g_bias + (long)m * n;
  float* mu_m = mu + (long)m * n;
  float* nu_m = nu + (long)m * n;
  const float bd = bias_decay[m];
  const float t = step_no[m];
  const float bc1 = 1.0f - powf(b1, t);
  const float bc2 = 1.0f - powf(b2, t);

  __shared__ float norm_sq_s;
  float decay_scale = 0.f;
  if (bd != 0.f) {
    float acc = 0.f;
    for (int j = threadIdx.x; j < n; j += NTHREADS) acc += b_m[j] * b_m[j];
    acc = wave_reduce_sum(acc);
    __shared__ float partial[4];
    if ((threadIdx.x & (WAVE - 1)) == 0) partial[threadIdx.x / WAVE] = acc;
    __syncthreads();
    if (threadIdx.x == 0) norm_sq_s = partial[0] + partial[1] + partial[2] + partial[3];
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
    b_m[j] -= lr * (m1 / bc1) / (sqrtf(v1 / bc2) + eps_adam);
  }
}
