// Hand-written CDNA4 (gfx950) kernels for spark_rapids_ml_amd.
//
// These replace the cuML/RAFT native layer of the reference:
// - kmeans_assign: fused pairwise-distance (MFMA f32) + argmin, the KMeansMG
//   Lloyd-step hot kernel (reference clustering.py:381-415 call site).
// - gram_f32: X^T X partials on MFMA f32 — PCAMG covariance
//   (feature.py:232-253) and LinearRegressionMG normal equations
//   (regression.py:549) building block.
// - softmax_residual_loss: fused softmax/sigmoid + residual + loss for the
//   LogisticRegressionMG L-BFGS iteration (classification.py:1046-1081).
// - label_accumulate: per-center sum/count scatter for the Lloyd update.
//
// MFMA notes (gfx950): v_mfma_f32_32x32x2_f32 — exact f32 at the 157 TF
// vector rate; lane l supplies A[i=l&31][k=l>>5] and B[k=l>>5][j=l&31];
// C/D element (reg, lane) -> row=(reg&3)+8*(reg>>2)+4*(lane>>5), col=lane&31.
// Wave = 64 lanes; LDS padded to stride 33 floats so 32-lane column reads are
// bank-conflict-free.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <cstdint>

#define HIP_CHECK(cmd)                                                         \
  do {                                                                         \
    hipError_t e = (cmd);                                                      \
    TORCH_CHECK(e == hipSuccess, "HIP error: ", hipGetErrorString(e));         \
  } while (0)

using f32x16 = __attribute__((ext_vector_type(16))) float;

// ---------------------------------------------------------------------------
// kmeans_assign: fused X·C^T (MFMA) + ||x||²+||c||²−2x·c + running argmin.
//
// Grid: one block per 128-row slice of X. Each block loops over all center
// tiles (BN=128) and, per tile, over d in BK=32 steps with LDS-staged X and C
// tiles. 4 waves, each computing a 2x2 grid of 32x32 MFMA tiles. The per-row
// running (dist,center) is kept packed in a 64-bit LDS word (float-bits<<32 |
// center) updated with atomicMin — float bits of non-negative distances sort
// correctly as unsigned.
// ---------------------------------------------------------------------------

constexpr int KM_BM = 128;
constexpr int KM_BN = 128;
constexpr int KM_BK = 64;
constexpr int KM_LD = KM_BK * KM_BM / 256;  // staging elems per thread (32)

__global__ __launch_bounds__(256) void kmeans_assign_kernel(
    const float* __restrict__ X,     // [n,d] row-major
    const float* __restrict__ C,     // [k,d] row-major
    const float* __restrict__ x_sq,  // [n]
    const float* __restrict__ c_sq,  // [k]
    int n, int d, int k,
    int32_t* __restrict__ labels,    // [n]
    float* __restrict__ min_dists,   // [n] squared distance to the winner
    double* __restrict__ inertia) {  // [1] accumulated
  // Single-buffer BK=64 with write-after-barrier register pipelining (T14):
  // per K-step 128 MFMAs (8192 issue cycles/wave) run between two barriers;
  // the next tile's registers are written right after the first barrier and
  // the tile after that's global loads are issued immediately, so HBM
  // latency hides under the MFMA phase. 2 blocks/CU (67 KB LDS).
  __shared__ float lds_x[KM_BK][KM_BM + 1];
  __shared__ float lds_c[KM_BK][KM_BN + 1];
  __shared__ unsigned long long best[KM_BM];

  const int i0 = blockIdx.x * KM_BM;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;  // wave row (0..1) -> 64 rows
  const int wc = wave & 1;   // wave col (0..1) -> 64 cols

  for (int i = tid; i < KM_BM; i += blockDim.x) best[i] = ~0ULL;

  const bool full_rows = (i0 + KM_BM <= n);
  const int nsteps = (d + KM_BK - 1) / KM_BK;
  float rx[KM_LD], rc[KM_LD];

  for (int j0 = 0; j0 < k; j0 += KM_BN) {
    const bool full_cols = (j0 + KM_BN <= k);
    f32x16 acc[2][2];
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int nn = 0; nn < 2; ++nn)
#pragma unroll
        for (int r = 0; r < 16; ++r) acc[m][nn][r] = 0.0f;

    // loads: KM_LD=32 elems/thread; e = q*256+tid over 8192 = 128 rows x 64 kd
#define KM_LOAD_T(dst, SRC, base_row, d0)                                      \
  do {                                                                         \
    if (full_rows && full_cols && (d0) + KM_BK <= d) {                         \
      _Pragma("unroll") for (int q = 0; q < KM_LD; ++q) {                      \
        int e = q * 256 + tid;                                                 \
        dst[q] = SRC[(int64_t)((base_row) + (e >> 6)) * d + (d0) + (e & 63)];  \
      }                                                                        \
    } else {                                                                   \
      _Pragma("unroll") for (int q = 0; q < KM_LD; ++q) {                      \
        int e = q * 256 + tid;                                                 \
        int gr = (base_row) + (e >> 6);                                        \
        int gd = (d0) + (e & 63);                                              \
        int lim = (&dst[0] == &rx[0]) ? n : k;                                 \
        dst[q] = (gr < lim && gd < d) ? SRC[(int64_t)gr * d + gd] : 0.0f;      \
      }                                                                        \
    }                                                                          \
  } while (0)

#define KM_WRITE()                                                             \
  do {                                                                         \
    _Pragma("unroll") for (int q = 0; q < KM_LD; ++q) {                        \
      int e = q * 256 + tid;                                                   \
      lds_x[e & 63][e >> 6] = rx[q];                                           \
      lds_c[e & 63][e >> 6] = rc[q];                                           \
    }                                                                          \
  } while (0)

    // prologue: tile 0 into LDS, tile 1 into regs
    KM_LOAD_T(rx, X, i0, 0);
    KM_LOAD_T(rc, C, j0, 0);
    __syncthreads();  // best[] init / previous epilogue complete
    KM_WRITE();
    if (nsteps > 1) {
      KM_LOAD_T(rx, X, i0, KM_BK);
      KM_LOAD_T(rc, C, j0, KM_BK);
    }
    __syncthreads();

    for (int step = 0; step < nsteps; ++step) {
#pragma unroll 8
      for (int kk = 0; kk < KM_BK / 2; ++kk) {
        const int kd = 2 * kk + (lane >> 5);
        float a0 = lds_x[kd][wr * 64 + (lane & 31)];
        float a1 = lds_x[kd][wr * 64 + 32 + (lane & 31)];
        float b0 = lds_c[kd][wc * 64 + (lane & 31)];
        float b1 = lds_c[kd][wc * 64 + 32 + (lane & 31)];
        acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
        acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
        acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
        acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
      }
      __syncthreads();  // all waves done reading this tile
      if (step + 1 < nsteps) {
        KM_WRITE();  // vmcnt-waits for loads issued a full MFMA phase ago
        if (step + 2 < nsteps) {
          KM_LOAD_T(rx, X, i0, (step + 2) * KM_BK);
          KM_LOAD_T(rc, C, j0, (step + 2) * KM_BK);
        }
        __syncthreads();
      }
    }
#undef KM_LOAD_T
#undef KM_WRITE

    // epilogue: distances + packed argmin into LDS
#pragma unroll
    for (int m = 0; m < 2; ++m) {
#pragma unroll
      for (int nn = 0; nn < 2; ++nn) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int row = wr * 64 + m * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
          int col = wc * 64 + nn * 32 + (lane & 31);
          int gi = i0 + row, gj = j0 + col;
          if (gi < n && gj < k) {
            float dist = x_sq[gi] + c_sq[gj] - 2.0f * acc[m][nn][r];
            dist = dist < 0.0f ? 0.0f : dist;
            unsigned long long packed =
                ((unsigned long long)__float_as_uint(dist) << 32) |
                (unsigned int)gj;
            atomicMin(&best[row], packed);
          }
        }
      }
    }
    __syncthreads();
  }

  // write labels / min dists; block-reduce inertia
  __shared__ double block_inertia[4];
  double partial = 0.0;
  for (int i = tid; i < KM_BM; i += blockDim.x) {
    int gi = i0 + i;
    if (gi < n) {
      unsigned long long p = best[i];
      float dist = __uint_as_float((unsigned int)(p >> 32));
      labels[gi] = (int32_t)(p & 0xffffffffu);
      min_dists[gi] = dist;
      partial += (double)dist;
    }
  }
  // wave reduce
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    partial += __shfl_down(partial, off, 64);
  if (lane == 0) block_inertia[wave] = partial;
  __syncthreads();
  if (tid == 0) {
    double s = block_inertia[0] + block_inertia[1] + block_inertia[2] + block_inertia[3];
    atomicAdd(inertia, s);
  }
}





// ---------------------------------------------------------------------------
// dbscan_sweep: fused n x n eps-neighborhood pass for DBSCAN, reusing the
// kmeans_assign tile structure (rows = this rank's slice of the replicated
// dataset, "centers" = ALL rows, tiled by 128). The torch path materializes
// [chunk, n] masked int64 label tensors (~8 bytes x n per row per sweep of
// pure HBM traffic); here the distance never leaves registers and the
// epilogue reduces straight into a 128-entry LDS accumulator.
//   mode 0: out[i] = #{ j : d2(i,j) <= eps2 }            (core counting)
//   mode 1: out[i] = min{ labels[j] : core[j], d2 <= eps2 }  (label sweep /
//           border assignment; 0x7fffffff when no core neighbor)
// One kernel serves core detection, the min-label propagation sweeps and
// the border pass (reference DBSCANMG: adjacency + BFS inside cuML,
// SURVEY.md §2.3b).
//
// Ball-cover pruning (reference algorithm="rbc", clustering.py:686-695):
// when tile_off != nullptr the column loop walks only the CSR list of
// admissible 128-column tiles for this 128-row block — tiles whose bounding
// ball can contain an eps-neighbor of the row block's ball (triangle
// inequality, computed host-side after a coarse-kmeans row permutation).
// The per-pair eps test below is unchanged, so results are exact.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void dbscan_sweep_kernel(
    const float* __restrict__ X,       // [n,d] replicated rows
    const float* __restrict__ x_sq,    // [n]
    int n, int d,
    int row0, int n_rows,              // this rank's slice [row0, row0+n_rows)
    float eps2, int mode,
    const uint8_t* __restrict__ core,  // [n] (mode 1)
    const int32_t* __restrict__ labels,// [n] (mode 1)
    const int32_t* __restrict__ tile_idx, // CSR col-tile ids (nullptr = dense)
    const int32_t* __restrict__ tile_off, // [n_row_blocks+1] (nullptr = dense)
    int32_t* __restrict__ out) {       // [n_rows]
  __shared__ float lds_x[KM_BK][KM_BM + 1];
  __shared__ float lds_c[KM_BK][KM_BN + 1];
  __shared__ int acc_row[KM_BM];

  const int i0 = row0 + blockIdx.x * KM_BM;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;
  const int wc = wave & 1;
  const int row_lim = row0 + n_rows;

  for (int i = tid; i < KM_BM; i += blockDim.x)
    acc_row[i] = (mode == 0) ? 0 : 0x7fffffff;

  const bool full_rows = (i0 + KM_BM <= row_lim);
  const int nsteps = (d + KM_BK - 1) / KM_BK;
  float rx[KM_LD], rc[KM_LD];
  // per-lane per-fragment-row accumulator carried across ALL column tiles:
  // dense data fires an LDS atomic per in-eps PAIR otherwise (~5e10 atomics
  // on 20-blob 1M x 64 — the label sweep cost 2.6x the count pass)
  int racc[2][16];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int r = 0; r < 16; ++r) racc[m][r] = (mode == 0) ? 0 : 0x7fffffff;

  const bool pruned = (tile_off != nullptr);
  const int t_beg = pruned ? tile_off[blockIdx.x] : 0;
  const int t_end = pruned ? tile_off[blockIdx.x + 1] : (n + KM_BN - 1) / KM_BN;
  for (int t = t_beg; t < t_end; ++t) {
    const int j0 = (pruned ? tile_idx[t] : t) * KM_BN;
    const bool full_cols = (j0 + KM_BN <= n);
    f32x16 acc[2][2];
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int nn = 0; nn < 2; ++nn)
#pragma unroll
        for (int r = 0; r < 16; ++r) acc[m][nn][r] = 0.0f;

#define DB_LOAD_T(dst, base_row, lim, d0)                                      \
  do {                                                                         \
    if (full_rows && full_cols && (d0) + KM_BK <= d) {                         \
      _Pragma("unroll") for (int q = 0; q < KM_LD; ++q) {                      \
        int e = q * 256 + tid;                                                 \
        dst[q] = X[(int64_t)((base_row) + (e >> 6)) * d + (d0) + (e & 63)];    \
      }                                                                        \
    } else {                                                                   \
      _Pragma("unroll") for (int q = 0; q < KM_LD; ++q) {                      \
        int e = q * 256 + tid;                                                 \
        int gr = (base_row) + (e >> 6);                                        \
        int gd = (d0) + (e & 63);                                              \
        dst[q] = (gr < (lim) && gd < d) ? X[(int64_t)gr * d + gd] : 0.0f;      \
      }                                                                        \
    }                                                                          \
  } while (0)

#define DB_WRITE()                                                             \
  do {                                                                         \
    _Pragma("unroll") for (int q = 0; q < KM_LD; ++q) {                        \
      int e = q * 256 + tid;                                                   \
      lds_x[e & 63][e >> 6] = rx[q];                                           \
      lds_c[e & 63][e >> 6] = rc[q];                                           \
    }                                                                          \
  } while (0)

    DB_LOAD_T(rx, i0, row_lim, 0);
    DB_LOAD_T(rc, j0, n, 0);
    __syncthreads();
    DB_WRITE();
    if (nsteps > 1) {
      DB_LOAD_T(rx, i0, row_lim, KM_BK);
      DB_LOAD_T(rc, j0, n, KM_BK);
    }
    __syncthreads();

    for (int step = 0; step < nsteps; ++step) {
#pragma unroll 8
      for (int kk = 0; kk < KM_BK / 2; ++kk) {
        const int kd = 2 * kk + (lane >> 5);
        float a0 = lds_x[kd][wr * 64 + (lane & 31)];
        float a1 = lds_x[kd][wr * 64 + 32 + (lane & 31)];
        float b0 = lds_c[kd][wc * 64 + (lane & 31)];
        float b1 = lds_c[kd][wc * 64 + 32 + (lane & 31)];
        acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
        acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
        acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
        acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
      }
      __syncthreads();
      if (step + 1 < nsteps) {
        DB_WRITE();
        if (step + 2 < nsteps) {
          DB_LOAD_T(rx, i0, row_lim, (step + 2) * KM_BK);
          DB_LOAD_T(rc, j0, n, (step + 2) * KM_BK);
        }
        __syncthreads();
      }
    }
#undef DB_LOAD_T
#undef DB_WRITE

    // epilogue: per-lane only 2 distinct columns -> hoist the per-column
    // gathers (x_sq / core / labels) out of the 32-row loop
#pragma unroll
    for (int nn = 0; nn < 2; ++nn) {
      const int col = wc * 64 + nn * 32 + (lane & 31);
      const int gj = j0 + col;
      if (gj >= n) continue;
      const float cs = x_sq[gj];
      int lab_j = 0;
      bool core_j = true;
      if (mode == 1) {
        core_j = core[gj] != 0;
        if (core_j) lab_j = labels[gj];
      }
      if (mode == 1 && !core_j) continue;
#pragma unroll
      for (int m = 0; m < 2; ++m) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int row = wr * 64 + m * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
          int gi = i0 + row;
          if (gi < row_lim) {
            float d2 = x_sq[gi] + cs - 2.0f * acc[m][nn][r];
            if (d2 <= eps2) {
              if (mode == 0)
                ++racc[m][r];
              else
                racc[m][r] = min(racc[m][r], lab_j);
            }
          }
        }
      }
    }
    __syncthreads();
  }

  // one LDS atomic per fragment row for the whole kernel
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int row = wr * 64 + m * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
      if (mode == 0) {
        if (racc[m][r]) atomicAdd(&acc_row[row], racc[m][r]);
      } else if (racc[m][r] != 0x7fffffff) {
        atomicMin(&acc_row[row], racc[m][r]);
      }
    }
  __syncthreads();

  for (int i = tid; i < KM_BM; i += blockDim.x) {
    int gi = i0 + i;
    if (gi < row_lim) out[gi - row0] = acc_row[i];
  }
}


// ---------------------------------------------------------------------------
// UMAP edge-sampled SGD (reference: cuML UMAP optimize_layout,
// SURVEY.md §2.3b umap fit). One thread per edge per epoch, Hogwild
// atomicAdd updates (the torch path's index_add is the same semantics with
// ~10 kernel launches + intermediate tensors per epoch; here one launch per
// epoch and the epoch loop lives in the host wrapper). Negative samples use
// a per-(edge, epoch, trial) wang-hash — distributionally equivalent to the
// torch generator, not bit-identical (UMAP is stochastic either way).
// ---------------------------------------------------------------------------

__device__ __forceinline__ unsigned umap_hash(unsigned x) {
  x = (x ^ 61u) ^ (x >> 16);
  x *= 9u;
  x = x ^ (x >> 4);
  x *= 0x27d4eb2du;
  x = x ^ (x >> 15);
  return x;
}

template <int DIM>
__global__ __launch_bounds__(256) void umap_sgd_epoch_kernel(
    float* __restrict__ emb,            // [n_head, DIM] updated in place
    float* __restrict__ tail_emb,       // [n_vertices, DIM] (== emb when fitting)
    const int32_t* __restrict__ heads,  // [m]
    const int32_t* __restrict__ tails,  // [m]
    const float* __restrict__ eps,      // [m] epochs-per-sample
    float* __restrict__ next_due,       // [m] state
    int m, int n_vertices,
    float a, float b, float alpha, float repulsion,
    int neg_rate, int move_tail, int epoch, unsigned seed) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= m) return;
  if (next_due[i] > (float)epoch) return;
  next_due[i] += eps[i];

  const float clip = 4.0f;
  const int h = heads[i], t = tails[i];
  // head update accumulates in REGISTERS across the positive edge and all
  // negative samples (sequential-local semantics, like umap-learn's CPU
  // loop), committed with ONE atomic per component at the end — the
  // per-sample version fired 2+2*neg_rate head atomics per due edge and the
  // atomic pipe (not bandwidth) bounded the epoch.
  float eh[DIM], diff[DIM], upd[DIM];
  float d2 = 0.0f;
#pragma unroll
  for (int c = 0; c < DIM; ++c) {
    eh[c] = emb[(int64_t)h * DIM + c];
    upd[c] = 0.0f;
    diff[c] = eh[c] - tail_emb[(int64_t)t * DIM + c];
    d2 += diff[c] * diff[c];
  }
  float d2c = fmaxf(d2, 1e-12f);
  float pb = __powf(d2c, b);
  float gcoef = (-2.0f * a * b * pb / d2c) / (1.0f + a * pb);
#pragma unroll
  for (int c = 0; c < DIM; ++c) {
    float g = fminf(fmaxf(gcoef * diff[c], -clip), clip);
    upd[c] += g;
    if (move_tail) atomicAdd(&tail_emb[(int64_t)t * DIM + c], -alpha * g);
  }

  unsigned rng = umap_hash(seed ^ umap_hash((unsigned)i * 2654435761u + (unsigned)epoch));
  for (int r = 0; r < neg_rate; ++r) {
    rng = umap_hash(rng + 0x9e3779b9u + (unsigned)r);
    int j = (int)(rng % (unsigned)n_vertices);
    d2 = 0.0f;
#pragma unroll
    for (int c = 0; c < DIM; ++c) {
      diff[c] = (eh[c] + alpha * upd[c]) - tail_emb[(int64_t)j * DIM + c];
      d2 += diff[c] * diff[c];
    }
    d2c = fmaxf(d2, 1e-12f);
    pb = __powf(d2c, b);
    gcoef = (2.0f * repulsion * b) / ((0.001f + d2) * (1.0f + a * pb));
#pragma unroll
    for (int c = 0; c < DIM; ++c) {
      float g = fminf(fmaxf(gcoef * diff[c], -clip), clip);
      upd[c] += g;
    }
  }
#pragma unroll
  for (int c = 0; c < DIM; ++c)
    if (upd[c] != 0.0f) atomicAdd(&emb[(int64_t)h * DIM + c], alpha * upd[c]);
}

// ---------------------------------------------------------------------------
// kmeans_assign_glds: same contract as kmeans_assign_kernel, staging via
// async global_load_lds DMA (16B) into a 16B-XOR-swizzled linear LDS image.
// 2 buffers, raw s_barrier + counted vmcnt (a plain __syncthreads would
// drain the in-flight next tile, guide §5 pipelining note). Requires d%4==0
// (16B-aligned rows); host falls back to the register-staged kernel else.
// All LDS lives in ONE __shared__ object (guide §5.4 trap (a)).
// ---------------------------------------------------------------------------

constexpr int KG_BM = 128;
constexpr int KG_BN = 128;
constexpr int KG_BK = 32;
// float offsets into the single LDS arena
constexpr int KG_X0 = 0;                    // lds_x buf0 [128*32]
constexpr int KG_X1 = 4096;                 // lds_x buf1
constexpr int KG_C0 = 8192;                 // lds_c buf0
constexpr int KG_C1 = 12288;                // lds_c buf1
constexpr int KG_BEST = 16384;              // u64 best[128] = 256 floats
constexpr int KG_SCRATCH = 16640;           // 8 doubles = 16 floats
constexpr int KG_TOTAL = 16656;

__device__ __forceinline__ int kg_swz(int i, int kd) {
  // element (row i, col kd) -> float offset within a 128x32 tile image:
  // 16B groups XOR-swizzled by (i&7) so column-strided b32 fragment reads
  // spread over 8 bank quads (<=4-way conflict) while glds stays lane-linear
  return i * KG_BK + ((((kd >> 2) ^ (i & 7)) << 2) | (kd & 3));
}

// ---------------------------------------------------------------------------
// kmeans_assign_256: 256x256 tile via glds, one 512-thread block (8 waves).
// The 128x128 kernel is HBM-traffic-bound at scale: X re-streams once per
// centroid tile (k/128 passes) and C once per row tile (n/128 passes) --
// ~190 GB for 1M x 3000 x k=1000 vs ~39 ms of pure MFMA work. Doubling both
// tile dims halves BOTH traffic terms (~95 GB). Staging goes through
// global_load_lds (zero staging registers -- the register-staged 256 variant
// spilled 384 VGPRs) with the same double-buffered vmcnt state machine as
// kmeans_assign_glds_kernel; 4 x 256x32 tile images + best[] = 131 KB LDS,
// one block per CU, still 2 waves/SIMD.
// ---------------------------------------------------------------------------

constexpr int KG2_BM = 256;
constexpr int KG2_BN = 256;
constexpr int KG2_X0 = 0;        // [256*32] floats
constexpr int KG2_X1 = 8192;
constexpr int KG2_C0 = 16384;
constexpr int KG2_C1 = 24576;
constexpr int KG2_BEST = 32768;  // u64 best[256] = 512 floats
constexpr int KG2_SCRATCH = 33280;  // 8 doubles = 16 floats
constexpr int KG2_TOTAL = 33296;

__global__ __launch_bounds__(512) void kmeans_assign_256_kernel(
    const float* __restrict__ X, const float* __restrict__ C,
    const float* __restrict__ x_sq, const float* __restrict__ c_sq,
    int n, int d, int k,
    int32_t* __restrict__ labels, float* __restrict__ min_dists,
    double* __restrict__ inertia) {
  __shared__ __attribute__((aligned(16))) float smem[KG2_TOTAL];

  const int i0 = blockIdx.x * KG2_BM;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;  // 0..3 -> 64-row band
  const int wc = wave & 1;   // 0..1 -> 128-col band

  unsigned long long* best = reinterpret_cast<unsigned long long*>(&smem[KG2_BEST]);
  for (int i = tid; i < KG2_BM; i += blockDim.x) best[i] = ~0ULL;

  const bool full_rows = (i0 + KG2_BM <= n);
  const int nsteps = (d + KG_BK - 1) / KG_BK;
  bool full_rows_cols = full_rows;

  auto pair_full = [&](int step) {
    return full_rows_cols && (step + 1) * KG_BK <= d;
  };

  // per-wave glds issue: 4 instructions per operand per tile (8 waves x 4
  // instr x 8 rows x 32 cols = the full 256x32 image)
  auto issue_glds = [&](const float* __restrict__ src, int base_row, int d0,
                        int lds_off) {
#pragma unroll
    for (int qq = 0; qq < 4; ++qq) {
      const int q = wave * 4 + qq;
      const int i = q * 8 + (lane >> 3);
      const int g = (lane & 7) ^ (i & 7);
      const float* gp = src + (int64_t)(base_row + i) * d + d0 + (g << 2);
      auto lp = (__attribute__((address_space(3))) uint32_t*)(&smem[lds_off + q * 256]);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)gp, lp, 16, 0, 0);
    }
  };
  auto issue_scalar = [&](const float* __restrict__ src, int base_row, int lim,
                          int d0, int lds_off) {
    for (int e = tid; e < KG2_BM * KG_BK; e += 512) {
      int i = e >> 5, kd = e & 31;
      int gr = base_row + i, gd = d0 + kd;
      smem[lds_off + kg_swz(i, kd)] =
          (gr < lim && gd < d) ? src[(int64_t)gr * d + gd] : 0.0f;
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  };
  auto issue_tile = [&](int step, int xoff, int coff, int j0) {
    const int d0 = step * KG_BK;
    if (pair_full(step)) {
      issue_glds(X, i0, d0, xoff);
      issue_glds(C, j0, d0, coff);
    } else {
      issue_scalar(X, i0, n, d0, xoff);
      issue_scalar(C, j0, k, d0, coff);
    }
  };

  for (int j0 = 0; j0 < k; j0 += KG2_BN) {
    const bool full_cols = (j0 + KG2_BN <= k);
    full_rows_cols = full_rows && full_cols;
    f32x16 acc[2][4];
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int nn = 0; nn < 4; ++nn)
#pragma unroll
        for (int r = 0; r < 16; ++r) acc[m][nn][r] = 0.0f;

    issue_tile(0, KG2_X0, KG2_C0, j0);
    if (nsteps > 1) issue_tile(1, KG2_X1, KG2_C1, j0);
    if (pair_full(0)) {
      if (nsteps > 1 && pair_full(1)) {
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
    }
    __builtin_amdgcn_s_barrier();

    for (int step = 0; step < nsteps; ++step) {
      const int xb = (step & 1) ? KG2_X1 : KG2_X0;
      const int cb = (step & 1) ? KG2_C1 : KG2_C0;
      const int xi = wr * 64 + (lane & 31);
      const int ci = wc * 128 + (lane & 31);
#pragma unroll 8
      for (int kk = 0; kk < KG_BK / 2; ++kk) {
        const int kd = 2 * kk + (lane >> 5);
        float a0 = smem[xb + kg_swz(xi, kd)];
        float a1 = smem[xb + kg_swz(xi + 32, kd)];
        float b0 = smem[cb + kg_swz(ci, kd)];
        float b1 = smem[cb + kg_swz(ci + 32, kd)];
        float b2 = smem[cb + kg_swz(ci + 64, kd)];
        float b3 = smem[cb + kg_swz(ci + 96, kd)];
        acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
        acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
        acc[0][2] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b2, acc[0][2], 0, 0, 0);
        acc[0][3] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b3, acc[0][3], 0, 0, 0);
        acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
        acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
        acc[1][2] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b2, acc[1][2], 0, 0, 0);
        acc[1][3] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b3, acc[1][3], 0, 0, 0);
      }
      __builtin_amdgcn_s_barrier();  // everyone done reading buf[step&1]
      if (step + 2 < nsteps) {
        issue_tile(step + 2, xb, cb, j0);
        if (pair_full(step + 1)) {
          if (pair_full(step + 2)) {
            asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
          } else {
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
          }
        }
        __builtin_amdgcn_s_barrier();
      } else if (step + 1 < nsteps) {
        if (pair_full(step + 1)) {
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();
      }
    }

    // epilogue: distances + packed argmin (no glds outstanding)
#pragma unroll
    for (int m = 0; m < 2; ++m) {
#pragma unroll
      for (int nn = 0; nn < 4; ++nn) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int row = wr * 64 + m * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
          int col = wc * 128 + nn * 32 + (lane & 31);
          int gi = i0 + row, gj = j0 + col;
          if (gi < n && gj < k) {
            float dist = x_sq[gi] + c_sq[gj] - 2.0f * acc[m][nn][r];
            dist = dist < 0.0f ? 0.0f : dist;
            unsigned long long packed =
                ((unsigned long long)__float_as_uint(dist) << 32) |
                (unsigned int)gj;
            atomicMin(&best[row], packed);
          }
        }
      }
    }
    __syncthreads();
  }

  double* block_inertia = reinterpret_cast<double*>(&smem[KG2_SCRATCH]);
  double partial = 0.0;
  for (int i = tid; i < KG2_BM; i += blockDim.x) {
    int gi = i0 + i;
    if (gi < n) {
      unsigned long long p = best[i];
      float dist = __uint_as_float((unsigned int)(p >> 32));
      labels[gi] = (int32_t)(p & 0xffffffffu);
      min_dists[gi] = dist;
      partial += (double)dist;
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    partial += __shfl_down(partial, off, 64);
  if (lane == 0) block_inertia[wave] = partial;
  __syncthreads();
  if (tid == 0) {
    double sum = 0.0;
#pragma unroll
    for (int w = 0; w < 8; ++w) sum += block_inertia[w];
    atomicAdd(inertia, sum);
  }
}


__global__ __launch_bounds__(256) void kmeans_assign_glds_kernel(
    const float* __restrict__ X, const float* __restrict__ C,
    const float* __restrict__ x_sq, const float* __restrict__ c_sq,
    int n, int d, int k,
    int32_t* __restrict__ labels, float* __restrict__ min_dists,
    double* __restrict__ inertia) {
  __shared__ __attribute__((aligned(16))) float smem[KG_TOTAL];

  const int i0 = blockIdx.x * KG_BM;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;
  const int wc = wave & 1;

  unsigned long long* best = reinterpret_cast<unsigned long long*>(&smem[KG_BEST]);
  for (int i = tid; i < KG_BM; i += blockDim.x) best[i] = ~0ULL;

  const bool full_rows = (i0 + KG_BM <= n);
  const int nsteps = (d + KG_BK - 1) / KG_BK;
  bool full_rows_cols = full_rows;  // && full_cols, set per j-tile

  // Tile "pair fullness": both operands of a d-step go glds (8 instructions
  // per wave) or both scalar — keeps the vmcnt immediates branch-constant.
  auto pair_full = [&](int step) {
    return full_rows_cols && (step + 1) * KG_BK <= d;
  };

  // per-wave glds issue: 4 instructions per operand per tile
  // instruction q (0..15 across 4 waves) covers rows [q*8, q*8+8)
  auto issue_glds = [&](const float* __restrict__ src, int base_row, int d0,
                        int lds_off) {
#pragma unroll
    for (int qq = 0; qq < 4; ++qq) {
      const int q = wave * 4 + qq;
      const int i = q * 8 + (lane >> 3);
      const int g = (lane & 7) ^ (i & 7);
      const float* gp = src + (int64_t)(base_row + i) * d + d0 + (g << 2);
      auto lp = (__attribute__((address_space(3))) uint32_t*)(&smem[lds_off + q * 256]);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)gp, lp, 16, 0, 0);
    }
  };
  auto issue_scalar = [&](const float* __restrict__ src, int base_row, int lim,
                          int d0, int lds_off) {
    for (int e = tid; e < KG_BM * KG_BK; e += 256) {
      int i = e >> 5, kd = e & 31;
      int gr = base_row + i, gd = d0 + kd;
      smem[lds_off + kg_swz(i, kd)] =
          (gr < lim && gd < d) ? src[(int64_t)gr * d + gd] : 0.0f;
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  };
  auto issue_tile = [&](int step, int xoff, int coff, int j0) {
    const int d0 = step * KG_BK;
    if (pair_full(step)) {
      issue_glds(X, i0, d0, xoff);
      issue_glds(C, j0, d0, coff);
    } else {
      issue_scalar(X, i0, n, d0, xoff);
      issue_scalar(C, j0, k, d0, coff);
    }
  };

  for (int j0 = 0; j0 < k; j0 += KG_BN) {
    const bool full_cols = (j0 + KG_BN <= k);
    full_rows_cols = full_rows && full_cols;
    f32x16 acc[2][2];
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int nn = 0; nn < 2; ++nn)
#pragma unroll
        for (int r = 0; r < 16; ++r) acc[m][nn][r] = 0.0f;

    // prologue: tile0 + tile1 in flight; wait until only tile1's glds remain
    issue_tile(0, KG_X0, KG_C0, j0);
    if (nsteps > 1) issue_tile(1, KG_X1, KG_C1, j0);
    if (pair_full(0)) {
      if (nsteps > 1 && pair_full(1)) {
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
    }
    __builtin_amdgcn_s_barrier();

    for (int step = 0; step < nsteps; ++step) {
      const int xb = (step & 1) ? KG_X1 : KG_X0;
      const int cb = (step & 1) ? KG_C1 : KG_C0;
      const int xi = wr * 64 + (lane & 31);
      const int ci = wc * 64 + (lane & 31);
#pragma unroll 8
      for (int kk = 0; kk < KG_BK / 2; ++kk) {
        const int kd = 2 * kk + (lane >> 5);
        float a0 = smem[xb + kg_swz(xi, kd)];
        float a1 = smem[xb + kg_swz(xi + 32, kd)];
        float b0 = smem[cb + kg_swz(ci, kd)];
        float b1 = smem[cb + kg_swz(ci + 32, kd)];
        acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
        acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
        acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
        acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
      }
      __builtin_amdgcn_s_barrier();  // everyone done reading buf[step&1]
      if (step + 2 < nsteps) {
        issue_tile(step + 2, xb, cb, j0);
        // wait until only tile (step+2)'s glds remain -> (step+1) landed
        if (pair_full(step + 1)) {
          if (pair_full(step + 2)) {
            asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
          } else {
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
          }
        }
        __builtin_amdgcn_s_barrier();
      } else if (step + 1 < nsteps) {
        if (pair_full(step + 1)) {
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();
      }
    }

    // epilogue: distances + packed argmin (no glds outstanding)
#pragma unroll
    for (int m = 0; m < 2; ++m) {
#pragma unroll
      for (int nn = 0; nn < 2; ++nn) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int row = wr * 64 + m * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
          int col = wc * 64 + nn * 32 + (lane & 31);
          int gi = i0 + row, gj = j0 + col;
          if (gi < n && gj < k) {
            float dist = x_sq[gi] + c_sq[gj] - 2.0f * acc[m][nn][r];
            dist = dist < 0.0f ? 0.0f : dist;
            unsigned long long packed =
                ((unsigned long long)__float_as_uint(dist) << 32) |
                (unsigned int)gj;
            atomicMin(&best[row], packed);
          }
        }
      }
    }
    __syncthreads();
  }

  double* block_inertia = reinterpret_cast<double*>(&smem[KG_SCRATCH]);
  double partial = 0.0;
  for (int i = tid; i < KG_BM; i += blockDim.x) {
    int gi = i0 + i;
    if (gi < n) {
      unsigned long long p = best[i];
      float dist = __uint_as_float((unsigned int)(p >> 32));
      labels[gi] = (int32_t)(p & 0xffffffffu);
      min_dists[gi] = dist;
      partial += (double)dist;
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    partial += __shfl_down(partial, off, 64);
  if (lane == 0) block_inertia[wave] = partial;
  __syncthreads();
  if (tid == 0)
    atomicAdd(inertia, block_inertia[0] + block_inertia[1] + block_inertia[2] +
                           block_inertia[3]);
}

// ---------------------------------------------------------------------------
// kmeans_assign_glds3: 3-buffer glds span (guide: "+83%" row) — ONE block
// per CU (99 KB LDS), 2 tile-pairs in flight across each raw barrier with
// counted vmcnt. Requires d%4==0.
// ---------------------------------------------------------------------------

constexpr int K3_BUF = 4096;  // floats per operand buffer

__global__ __launch_bounds__(256) void kmeans_assign_glds3_kernel(
    const float* __restrict__ X, const float* __restrict__ C,
    const float* __restrict__ x_sq, const float* __restrict__ c_sq,
    int n, int d, int k,
    int32_t* __restrict__ labels, float* __restrict__ min_dists,
    double* __restrict__ inertia) {
  // arena: x bufs [0,3*4096), c bufs [3*4096, 6*4096), best, scratch
  __shared__ __attribute__((aligned(16))) float smem[6 * K3_BUF + 256 + 16];
  const int BEST = 6 * K3_BUF;
  const int SCR = BEST + 256;

  const int i0 = blockIdx.x * KG_BM;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;
  const int wc = wave & 1;

  unsigned long long* best = reinterpret_cast<unsigned long long*>(&smem[BEST]);
  for (int i = tid; i < KG_BM; i += blockDim.x) best[i] = ~0ULL;

  const bool full_rows = (i0 + KG_BM <= n);
  const int nsteps = (d + KG_BK - 1) / KG_BK;
  bool frc = full_rows;

  auto pf = [&](int step) { return frc && (step + 1) * KG_BK <= d; };

  auto issue_glds = [&](const float* __restrict__ src, int base_row, int d0,
                        int lds_off) {
#pragma unroll
    for (int qq = 0; qq < 4; ++qq) {
      const int q = wave * 4 + qq;
      const int i = q * 8 + (lane >> 3);
      const int g = (lane & 7) ^ (i & 7);
      const float* gp = src + (int64_t)(base_row + i) * d + d0 + (g << 2);
      auto lp = (__attribute__((address_space(3))) uint32_t*)(&smem[lds_off + q * 256]);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)gp, lp, 16, 0, 0);
    }
  };
  auto issue_scalar = [&](const float* __restrict__ src, int base_row, int lim,
                          int d0, int lds_off) {
    for (int e = tid; e < KG_BM * KG_BK; e += 256) {
      int i = e >> 5, kd = e & 31;
      int gr = base_row + i, gd = d0 + kd;
      smem[lds_off + kg_swz(i, kd)] =
          (gr < lim && gd < d) ? src[(int64_t)gr * d + gd] : 0.0f;
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  };
  auto issue_tile = [&](int step, int j0) {
    const int d0 = step * KG_BK;
    const int b = step % 3;
    if (pf(step)) {
      issue_glds(X, i0, d0, b * K3_BUF);
      issue_glds(C, j0, d0, (3 + b) * K3_BUF);
    } else {
      issue_scalar(X, i0, n, d0, b * K3_BUF);
      issue_scalar(C, j0, k, d0, (3 + b) * K3_BUF);
    }
  };
  // wait until only tiles {a, b} (8 glds each when pair-full) remain
  auto wait_leaving = [&](int ta, int tb) {
    const bool fa = ta >= 0 && ta < nsteps && pf(ta);
    const bool fb = tb >= 0 && tb < nsteps && pf(tb);
    if (fa && fb) {
      asm volatile("s_waitcnt vmcnt(16)" ::: "memory");
    } else if (fa || fb) {
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
  };

  for (int j0 = 0; j0 < k; j0 += KG_BN) {
    frc = full_rows && (j0 + KG_BN <= k);
    f32x16 acc[2][2];
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int nn = 0; nn < 2; ++nn)
#pragma unroll
        for (int r = 0; r < 16; ++r) acc[m][nn][r] = 0.0f;

    issue_tile(0, j0);
    if (nsteps > 1) issue_tile(1, j0);
    if (nsteps > 2) issue_tile(2, j0);
    wait_leaving(nsteps > 1 ? 1 : -1, nsteps > 2 ? 2 : -1);  // tile0 landed
    __builtin_amdgcn_s_barrier();

    for (int step = 0; step < nsteps; ++step) {
      const int xb = (step % 3) * K3_BUF;
      const int cb = (3 + step % 3) * K3_BUF;
      const int xi = wr * 64 + (lane & 31);
      const int ci = wc * 64 + (lane & 31);
#pragma unroll 8
      for (int kk = 0; kk < KG_BK / 2; ++kk) {
        const int kd = 2 * kk + (lane >> 5);
        float a0 = smem[xb + kg_swz(xi, kd)];
        float a1 = smem[xb + kg_swz(xi + 32, kd)];
        float b0 = smem[cb + kg_swz(ci, kd)];
        float b1 = smem[cb + kg_swz(ci + 32, kd)];
        acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
        acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
        acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
        acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
      }
      __builtin_amdgcn_s_barrier();  // all waves done reading buf[step%3]
      if (step + 3 < nsteps) issue_tile(step + 3, j0);
      if (step + 1 < nsteps) {
        wait_leaving(step + 2, step + 3);  // (step+1) landed
        __builtin_amdgcn_s_barrier();
      }
    }

#pragma unroll
    for (int m = 0; m < 2; ++m) {
#pragma unroll
      for (int nn = 0; nn < 2; ++nn) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int row = wr * 64 + m * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
          int col = wc * 64 + nn * 32 + (lane & 31);
          int gi = i0 + row, gj = j0 + col;
          if (gi < n && gj < k) {
            float dist = x_sq[gi] + c_sq[gj] - 2.0f * acc[m][nn][r];
            dist = dist < 0.0f ? 0.0f : dist;
            unsigned long long packed =
                ((unsigned long long)__float_as_uint(dist) << 32) |
                (unsigned int)gj;
            atomicMin(&best[row], packed);
          }
        }
      }
    }
    __syncthreads();
  }

  double* block_inertia = reinterpret_cast<double*>(&smem[SCR]);
  double partial = 0.0;
  for (int i = tid; i < KG_BM; i += blockDim.x) {
    int gi = i0 + i;
    if (gi < n) {
      unsigned long long p = best[i];
      float dist = __uint_as_float((unsigned int)(p >> 32));
      labels[gi] = (int32_t)(p & 0xffffffffu);
      min_dists[gi] = dist;
      partial += (double)dist;
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    partial += __shfl_down(partial, off, 64);
  if (lane == 0) block_inertia[wave] = partial;
  __syncthreads();
  if (tid == 0)
    atomicAdd(inertia, block_inertia[0] + block_inertia[1] + block_inertia[2] +
                           block_inertia[3]);
}

// ---------------------------------------------------------------------------
// kmeans_assign_sb3: BM=64 x BN=256, BK=32, SINGLE-buffer glds staging.
// 41 KB LDS -> 3 blocks/CU (3 waves/SIMD cover each other's staging) and
// X is re-streamed only k/256 times. Requires d%4==0.
// ---------------------------------------------------------------------------

constexpr int KS_BM = 64;
constexpr int KS_BN = 256;
constexpr int KS_BK = 32;
constexpr int KS_X0 = 0;                       // 64*32 floats
constexpr int KS_C0 = KS_BM * KS_BK;           // 256*32 floats
constexpr int KS_BEST = KS_C0 + KS_BN * KS_BK; // u64[64] = 128 floats
constexpr int KS_SCR = KS_BEST + 128;
constexpr int KS_TOTAL = KS_SCR + 16;

__global__ __launch_bounds__(256) void kmeans_assign_sb3_kernel(
    const float* __restrict__ X, const float* __restrict__ C,
    const float* __restrict__ x_sq, const float* __restrict__ c_sq,
    int n, int d, int k,
    int32_t* __restrict__ labels, float* __restrict__ min_dists,
    double* __restrict__ inertia) {
  __shared__ __attribute__((aligned(16))) float smem[KS_TOTAL];

  const int i0 = blockIdx.x * KS_BM;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;   // wave = column quarter (wc)

  unsigned long long* best = reinterpret_cast<unsigned long long*>(&smem[KS_BEST]);
  for (int i = tid; i < KS_BM; i += blockDim.x) best[i] = ~0ULL;

  const bool full_rows = (i0 + KS_BM <= n);
  const int nsteps = (d + KS_BK - 1) / KS_BK;

  // glds: instruction q covers rows [q*8, q*8+8) of a [rows][32]-float image
  auto issue_glds = [&](const float* __restrict__ src, int base_row, int d0,
                        int lds_off, int n_instr_per_wave) {
    for (int qq = 0; qq < n_instr_per_wave; ++qq) {
      const int q = wave * n_instr_per_wave + qq;
      const int i = q * 8 + (lane >> 3);
      const int g = (lane & 7) ^ (i & 7);
      const float* gp = src + (int64_t)(base_row + i) * d + d0 + (g << 2);
      auto lp = (__attribute__((address_space(3))) uint32_t*)(&smem[lds_off + q * 256]);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)gp, lp, 16, 0, 0);
    }
  };
  auto issue_scalar = [&](const float* __restrict__ src, int base_row, int lim,
                          int d0, int lds_off, int rows) {
    for (int e = tid; e < rows * KS_BK; e += 256) {
      int i = e >> 5, kd = e & 31;
      int gr = base_row + i, gd = d0 + kd;
      smem[lds_off + kg_swz(i, kd)] =
          (gr < lim && gd < d) ? src[(int64_t)gr * d + gd] : 0.0f;
    }
  };

  for (int j0 = 0; j0 < k; j0 += KS_BN) {
    const bool pair_full_cols = full_rows && (j0 + KS_BN <= k);
    f32x16 acc[2][2];
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int nn = 0; nn < 2; ++nn)
#pragma unroll
        for (int r = 0; r < 16; ++r) acc[m][nn][r] = 0.0f;

    for (int step = 0; step < nsteps; ++step) {
      const int d0 = step * KS_BK;
      __syncthreads();  // previous MFMA reads done (and best[] init)
      if (pair_full_cols && d0 + KS_BK <= d) {
        issue_glds(X, i0, d0, KS_X0, 2);
        issue_glds(C, j0, d0, KS_C0, 8);
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      } else {
        issue_scalar(X, i0, n, d0, KS_X0, KS_BM);
        issue_scalar(C, j0, k, d0, KS_C0, KS_BN);
      }
      __syncthreads();

      const int ci = wave * 64 + (lane & 31);
#pragma unroll 8
      for (int kk = 0; kk < KS_BK / 2; ++kk) {
        const int kd = 2 * kk + (lane >> 5);
        float a0 = smem[KS_X0 + kg_swz(lane & 31, kd)];
        float a1 = smem[KS_X0 + kg_swz((lane & 31) + 32, kd)];
        float b0 = smem[KS_C0 + kg_swz(ci, kd)];
        float b1 = smem[KS_C0 + kg_swz(ci + 32, kd)];
        acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
        acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
        acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
        acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
      }
    }
    __syncthreads();

#pragma unroll
    for (int m = 0; m < 2; ++m) {
#pragma unroll
      for (int nn = 0; nn < 2; ++nn) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int row = m * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
          int col = wave * 64 + nn * 32 + (lane & 31);
          int gi = i0 + row, gj = j0 + col;
          if (gi < n && gj < k) {
            float dist = x_sq[gi] + c_sq[gj] - 2.0f * acc[m][nn][r];
            dist = dist < 0.0f ? 0.0f : dist;
            unsigned long long packed =
                ((unsigned long long)__float_as_uint(dist) << 32) |
                (unsigned int)gj;
            atomicMin(&best[row], packed);
          }
        }
      }
    }
    __syncthreads();
  }

  double* block_inertia = reinterpret_cast<double*>(&smem[KS_SCR]);
  double partial = 0.0;
  for (int i = tid; i < KS_BM; i += blockDim.x) {
    int gi = i0 + i;
    if (gi < n) {
      unsigned long long p = best[i];
      float dist = __uint_as_float((unsigned int)(p >> 32));
      labels[gi] = (int32_t)(p & 0xffffffffu);
      min_dists[gi] = dist;
      partial += (double)dist;
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    partial += __shfl_down(partial, off, 64);
  if (lane == 0) block_inertia[wave] = partial;
  __syncthreads();
  if (tid == 0)
    atomicAdd(inertia, block_inertia[0] + block_inertia[1] + block_inertia[2] +
                           block_inertia[3]);
}

// ---------------------------------------------------------------------------
// label_accumulate: sums[label[i]] += X[i], counts[label[i]] += 1 via a
// sort-based segmented reduction: rows pre-sorted by label (torch.sort in
// the wrapper), each (label, split) block register-accumulates its column
// stripe over its row sub-segment — each X element is read ONCE, coalesced,
// and the only atomics are the k*SPLIT partial merges.
// ---------------------------------------------------------------------------

constexpr int LA_SPLIT = 8;    // sub-segments per label (skew tolerance)
constexpr int LA_COLS = 16;    // column accumulators per thread (256*16=4096)

__global__ __launch_bounds__(256) void segment_sum_kernel(
    const float* __restrict__ X, const int64_t* __restrict__ perm,
    const int64_t* __restrict__ seg_off, int d,
    float* __restrict__ sums, float* __restrict__ counts) {
  const int lab = blockIdx.x / LA_SPLIT;
  const int split = blockIdx.x % LA_SPLIT;
  const int64_t s0 = seg_off[lab], e0 = seg_off[lab + 1];
  const int64_t len = e0 - s0;
  if (len == 0) return;
  const int64_t chunk = (len + LA_SPLIT - 1) / LA_SPLIT;
  const int64_t rs = s0 + split * chunk;
  const int64_t re = min(e0, rs + chunk);
  if (rs >= re) return;
  if (split == 0 && threadIdx.x == 0) counts[lab] = (float)len;

  for (int cb = 0; cb < d; cb += blockDim.x * LA_COLS) {
    float acc[LA_COLS];
#pragma unroll
    for (int q = 0; q < LA_COLS; ++q) acc[q] = 0.0f;
    int64_t r = rs;
    for (; r + 4 <= re; r += 4) {  // 4 rows in flight for latency hiding
      const float* row0 = X + perm[r] * (int64_t)d;
      const float* row1 = X + perm[r + 1] * (int64_t)d;
      const float* row2 = X + perm[r + 2] * (int64_t)d;
      const float* row3 = X + perm[r + 3] * (int64_t)d;
#pragma unroll
      for (int q = 0; q < LA_COLS; ++q) {
        int c = cb + q * blockDim.x + threadIdx.x;
        if (c < d) acc[q] += row0[c] + row1[c] + row2[c] + row3[c];
      }
    }
    for (; r < re; ++r) {
      const float* row = X + perm[r] * (int64_t)d;
#pragma unroll
      for (int q = 0; q < LA_COLS; ++q) {
        int c = cb + q * blockDim.x + threadIdx.x;
        if (c < d) acc[q] += row[c];
      }
    }
#pragma unroll
    for (int q = 0; q < LA_COLS; ++q) {
      int c = cb + q * blockDim.x + threadIdx.x;
      if (c < d) {
        if (LA_SPLIT > 1)
          atomicAdd(&sums[(int64_t)lab * d + c], acc[q]);
        else
          sums[(int64_t)lab * d + c] = acc[q];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// label_accumulate_lds: direct one-pass scatter for small k*d. Each block
// owns an LDS-private [k,d] f32 accumulator (+ k counts), streams its row
// range in NATURAL order — X read once, coalesced, no sort and no permuted
// gather — and atomically merges into global once at the end (same
// privatization pattern as csr_grad). Used when k*d*4 <= 120 KB: on the
// BASELINE k=200 d=128 config the sort+segment path's permuted row gather
// ran at ~0.4 TB/s (128 ms for 51 GB); this pass is a plain streaming read.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(512) void label_accumulate_lds_kernel(
    const float* __restrict__ X, const int32_t* __restrict__ labels,
    int64_t n, int d, int k, int64_t rows_per_block,
    float* __restrict__ sums, float* __restrict__ counts) {
  extern __shared__ float lacc[];  // [k*d] sums then [k] counts
  float* lcnt = lacc + (size_t)k * d;
  const int kd = k * d;
  for (int i = threadIdx.x; i < kd + k; i += blockDim.x) lacc[i] = 0.0f;
  __syncthreads();
  const int64_t rs = (int64_t)blockIdx.x * rows_per_block;
  const int64_t re = min(n, rs + rows_per_block);
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int nw = (int)(blockDim.x >> 6);
  if ((d & 1) == 0 && d <= 128) {
    // float2 row reads (d=128 -> one 512 B wave access covers the row).
    // Two-deep software pipeline over 8-row groups: the ISA of the naive
    // form showed 8 SERIALIZED uniform label loads (staggered vmcnt(7..0))
    // and no cross-iteration overlap — ~2 full HBM latencies per group per
    // wave. Here the 8 labels arrive in ONE 8-lane load (broadcast by
    // shuffles) and the NEXT group's label+row loads are issued before the
    // current group's waits, so every address (a function of r alone)
    // pipelines.
    const int d2i = d >> 1;
    const bool act = lane < d2i;
    // each wave streams a CONTIGUOUS row slice: an 8-row group is one 4 KB
    // burst (the interleaved wave-stride layout issued 8 scattered 512 B
    // segments per group and capped at ~790 GB/s)
    const int64_t span = (re - rs + nw - 1) / nw;
    int64_t r = min(re, rs + (int64_t)wave * span);
    const int64_t we = min(re, r + span);
    int labv = (lane < 8 && r + lane < we) ? labels[r + lane] : 0;
    float2 va[8], vb[8];
#pragma unroll
    for (int q = 0; q < 8; ++q) {
      const int64_t rr = r + q;
      va[q] = (act && rr < we)
                  ? reinterpret_cast<const float2*>(X + rr * (int64_t)d)[lane]
                  : float2{0.0f, 0.0f};
    }
    for (; r + 7 < we;) {
      const int64_t rn = r + 8;
      int labn = 0;
      if (rn + 7 < we) {
        labn = (lane < 8) ? labels[rn + lane] : 0;
#pragma unroll
        for (int q = 0; q < 8; ++q)
          vb[q] = act ? reinterpret_cast<const float2*>(
                            X + (rn + q) * (int64_t)d)[lane]
                      : float2{0.0f, 0.0f};
      }
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        const int lq = __shfl(labv, q, 64);
        float* dst = lacc + (size_t)lq * d;
        if (act) {
          atomicAdd(&dst[2 * lane], va[q].x);
          atomicAdd(&dst[2 * lane + 1], va[q].y);
        }
        if (lane == 0) atomicAdd(&lcnt[lq], 1.0f);
      }
      r = rn;
      labv = labn;
#pragma unroll
      for (int q = 0; q < 8; ++q) va[q] = vb[q];
    }
    for (; r < we; ++r) {
      const float2* row = reinterpret_cast<const float2*>(X + r * (int64_t)d);
      float* dst = lacc + (size_t)labels[r] * d;
      for (int c2 = lane; c2 < d2i; c2 += 64) {
        float2 v = row[c2];
        atomicAdd(&dst[2 * c2], v.x);
        atomicAdd(&dst[2 * c2 + 1], v.y);
      }
      if (lane == 0) atomicAdd(&lcnt[labels[r]], 1.0f);
    }
  } else if ((d & 1) == 0) {
    int64_t r = rs + wave;
    for (; r < re; r += nw) {
      const float2* row = reinterpret_cast<const float2*>(X + r * (int64_t)d);
      float* dst = lacc + (size_t)labels[r] * d;
      for (int c2 = lane; c2 < (d >> 1); c2 += 64) {
        float2 v = row[c2];
        atomicAdd(&dst[2 * c2], v.x);
        atomicAdd(&dst[2 * c2 + 1], v.y);
      }
      if (lane == 0) atomicAdd(&lcnt[labels[r]], 1.0f);
    }
  } else {
    for (int64_t r = rs + wave; r < re; r += nw) {
      const float* row = X + r * (int64_t)d;
      float* dst = lacc + (size_t)labels[r] * d;
      for (int c = lane; c < d; c += 64) atomicAdd(&dst[c], row[c]);
      if (lane == 0) atomicAdd(&lcnt[labels[r]], 1.0f);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < kd; i += blockDim.x)
    if (lacc[i] != 0.0f) atomicAdd(&sums[i], lacc[i]);
  for (int i = threadIdx.x; i < k; i += blockDim.x)
    if (lcnt[i] != 0.0f) atomicAdd(&counts[i], lcnt[i]);
}

// ---------------------------------------------------------------------------
// gram_f32: C[d,d] = A^T A for A [n,d] row-major (TN MFMA GEMM).
// Upper blocks only would halve work; full matrix kept for simplicity —
// the symmetric skip is a planned optimization.
// ---------------------------------------------------------------------------

constexpr int GR_BM = 128;
constexpr int GR_BN = 128;
constexpr int GR_BK = 64;
constexpr int GR_LD = GR_BK * GR_BM / 256;  // 32 staging elems per thread

// gram_kernel: C[d,d] = A^T A, upper-triangle block enumeration (1D grid of
// nb*(nb+1)/2 tiles x SPLIT row-slices; partials atomicAdd'ed) with the
// same BK=64 write-after-barrier pipeline as the kmeans assign kernel.
__global__ __launch_bounds__(256) void gram_kernel(
    const float* __restrict__ A, int64_t n, int d, int nb, int split,
    float* __restrict__ out) {
  __shared__ float lds_i[GR_BK][GR_BM + 1];
  __shared__ float lds_j[GR_BK][GR_BN + 1];

  const int tri = blockIdx.x / split;
  const int slice = blockIdx.x % split;
  // tri -> (bi, bj) upper triangle: bi = row of the triangle
  int bi = (int)((sqrtf(8.0f * tri + 1.0f) - 1.0f) * 0.5f);
  while ((bi + 1) * (bi + 2) / 2 <= tri) ++bi;
  while (bi * (bi + 1) / 2 > tri) --bi;
  const int bj_off = tri - bi * (bi + 1) / 2;
  const int j0 = bi * GR_BN;          // bi-th diagonal stripe
  const int i0 = bj_off * GR_BM;      // column tile within the stripe (i0<=j0)

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;
  const int wc = wave & 1;

  const int nsteps_total = (int)((n + GR_BK - 1) / GR_BK);
  const int per = (nsteps_total + split - 1) / split;
  const int s_begin = slice * per;
  const int s_end = min(nsteps_total, s_begin + per);
  if (s_begin >= s_end) return;

  const bool full_i = (i0 + GR_BM <= d);
  const bool full_j = (j0 + GR_BN <= d);

  f32x16 acc[2][2];
#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int nn = 0; nn < 2; ++nn)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[m][nn][r] = 0.0f;

  float rg[GR_LD];

#define GR_LOAD(dst_unused, col0, step, full)                                  \
  do {                                                                         \
    int64_t r0 = (int64_t)(step)*GR_BK;                                        \
    if ((full) && r0 + GR_BK <= n) {                                           \
      _Pragma("unroll") for (int q = 0; q < GR_LD; ++q) {                      \
        int e = q * 256 + tid;                                                 \
        rg[q] = A[(r0 + (e >> 7)) * d + (col0) + (e & 127)];                   \
      }                                                                        \
    } else {                                                                   \
      _Pragma("unroll") for (int q = 0; q < GR_LD; ++q) {                      \
        int e = q * 256 + tid;                                                 \
        int64_t gr = r0 + (e >> 7);                                            \
        int gc = (col0) + (e & 127);                                           \
        rg[q] = (gr < n && gc < d) ? A[gr * d + gc] : 0.0f;                    \
      }                                                                        \
    }                                                                          \
  } while (0)

#define GR_WRITE(dstlds)                                                       \
  do {                                                                         \
    _Pragma("unroll") for (int q = 0; q < GR_LD; ++q) {                        \
      int e = q * 256 + tid;                                                   \
      dstlds[e >> 7][e & 127] = rg[q];                                         \
    }                                                                          \
  } while (0)

  // Staging: consecutive lanes take consecutive COLUMNS of one data row
  // (row-major A -> coalesced 512B reads); LDS write [rr][i] is contiguous.
  GR_LOAD(rg, i0, s_begin, full_i);
  GR_WRITE(lds_i);
  GR_LOAD(rg, j0, s_begin, full_j);
  GR_WRITE(lds_j);
  __syncthreads();

  for (int step = s_begin; step < s_end; ++step) {
#pragma unroll 8
    for (int kk = 0; kk < GR_BK / 2; ++kk) {
      const int rr = 2 * kk + (lane >> 5);
      float a0 = lds_i[rr][wr * 64 + (lane & 31)];
      float a1 = lds_i[rr][wr * 64 + 32 + (lane & 31)];
      float b0 = lds_j[rr][wc * 64 + (lane & 31)];
      float b1 = lds_j[rr][wc * 64 + 32 + (lane & 31)];
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
    if (step + 1 < s_end) {
      GR_LOAD(rg, i0, step + 1, full_i);
      GR_WRITE(lds_i);
      GR_LOAD(rg, j0, step + 1, full_j);
      GR_WRITE(lds_j);
      __syncthreads();
    }
  }
#undef GR_LOAD
#undef GR_WRITE

#pragma unroll
  for (int m = 0; m < 2; ++m)
#pragma unroll
    for (int nn = 0; nn < 2; ++nn)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int row = i0 + wr * 64 + m * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
        int col = j0 + wc * 64 + nn * 32 + (lane & 31);
        if (row < d && col < d && col >= row) {
          float v = acc[m][nn][r];
          if (split > 1) {
            atomicAdd(&out[(int64_t)row * d + col], v);
          } else {
            out[(int64_t)row * d + col] = v;
          }
        }
      }
}

// mirror upper triangle to lower
__global__ void gram_mirror_kernel(float* __restrict__ out, int d) {
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t total = (int64_t)d * d;
  for (; idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    int row = (int)(idx / d), col = (int)(idx % d);
    if (col < row) out[idx] = out[(int64_t)col * d + row];
  }
}

// ---------------------------------------------------------------------------
// knn_select: fused brute-force k-NN — MFMA distance tiles + in-LDS
// per-query top-k (reference NearestNeighborsMG's tiled ||x-y||^2 + top-k
// merge, SURVEY.md §2.3b knn.py:763-774). No [nq, ni] distance matrix is
// materialized: each 64-query block streams item tiles, keeps an unsorted
// k-candidate pool per query in LDS (one merge lane per query, current-max
// cached in registers), and writes its item-range partial top-k; partials
// across item splits merge on the host side (torch.topk over [q, S*k]).
// k <= 64.
// ---------------------------------------------------------------------------

constexpr int KN_QB = 64;    // queries per block
constexpr int KN_IB = 128;   // items per tile
constexpr int KN_BK = 32;    // d chunk
constexpr int KN_KMAX = 64;

__global__ __launch_bounds__(256) void knn_select_kernel(
    const float* __restrict__ Q,     // [nq, d]
    const float* __restrict__ I,     // [ni, d]
    const float* __restrict__ q_sq,  // [nq]
    const float* __restrict__ i_sq,  // [ni]
    int nq, int ni, int d, int k, int isplit,
    float* __restrict__ out_dist,    // [nq, isplit, k]
    int32_t* __restrict__ out_idx) { // [nq, isplit, k]
  // floats: lds_q [32][65], lds_i [32][129], dist [64][129], pool u64[64][k]
  __shared__ __attribute__((aligned(16))) float smem[32 * 65 + 32 * 129 + 64 * 129 +
                                                     2 * KN_QB * KN_KMAX];
  constexpr int SQ = 0;
  constexpr int SI = 32 * 65;
  constexpr int SD = SI + 32 * 129;
  constexpr int SP = SD + 64 * 129;

  const int q0 = (blockIdx.x / isplit) * KN_QB;
  const int split = blockIdx.x % isplit;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;  // 0..1 query half (32 rows)
  const int wc = wave & 1;   // 0..1 item half (64 cols)

  unsigned long long* pool = reinterpret_cast<unsigned long long*>(&smem[SP]);
  for (int e = tid; e < KN_QB * k; e += 256) pool[e] = ~0ULL;

  const int per = (ni + isplit - 1) / isplit;
  const int it_begin = split * per;
  const int it_end = min(ni, it_begin + per);

  // merge-lane registers: thread t (<64) owns query q0+t
  float cur_max = __uint_as_float(0x7f7fffffu);  // FLT_MAX
  int cur_pos = 0;
  int filled = 0;

  for (int ib = it_begin; ib < it_end; ib += KN_IB) {
    f32x16 acc[2];
#pragma unroll
    for (int nn = 0; nn < 2; ++nn)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[nn][r] = 0.0f;

    const bool tile_full = (q0 + KN_QB <= nq) && (ib + KN_IB <= it_end);
    for (int d0 = 0; d0 < d; d0 += KN_BK) {
      // stage Q[q0:q0+64, d0:+32] -> lds_q[kd][row], I tile likewise.
      // Guards hoisted to tile level (per-element selects de-pipeline
      // hipcc, guide §5.4 trap (c)).
      if (tile_full && d0 + KN_BK <= d) {
#pragma unroll 4
        for (int e = tid; e < KN_QB * KN_BK; e += 256) {
          int row = e >> 5, kd = e & 31;
          smem[SQ + kd * 65 + row] = Q[(int64_t)(q0 + row) * d + d0 + kd];
        }
#pragma unroll 4
        for (int e = tid; e < KN_IB * KN_BK; e += 256) {
          int row = e >> 5, kd = e & 31;
          smem[SI + kd * 129 + row] = I[(int64_t)(ib + row) * d + d0 + kd];
        }
      } else {
        for (int e = tid; e < KN_QB * KN_BK; e += 256) {
          int row = e >> 5, kd = e & 31;
          int gq = q0 + row, gd = d0 + kd;
          smem[SQ + kd * 65 + row] =
              (gq < nq && gd < d) ? Q[(int64_t)gq * d + gd] : 0.0f;
        }
        for (int e = tid; e < KN_IB * KN_BK; e += 256) {
          int row = e >> 5, kd = e & 31;
          int gi = ib + row, gd = d0 + kd;
          smem[SI + kd * 129 + row] =
              (gi < it_end && gd < d) ? I[(int64_t)gi * d + gd] : 0.0f;
        }
      }
      __syncthreads();
#pragma unroll 8
      for (int kk = 0; kk < KN_BK / 2; ++kk) {
        const int kd = 2 * kk + (lane >> 5);
        float a = smem[SQ + kd * 65 + wr * 32 + (lane & 31)];
        float b0 = smem[SI + kd * 129 + wc * 64 + (lane & 31)];
        float b1 = smem[SI + kd * 129 + wc * 64 + 32 + (lane & 31)];
        acc[0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b0, acc[0], 0, 0, 0);
        acc[1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b1, acc[1], 0, 0, 0);
      }
      __syncthreads();
    }

    // distances into SD[row][col]
#pragma unroll
    for (int nn = 0; nn < 2; ++nn) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int row = wr * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
        int col = wc * 64 + nn * 32 + (lane & 31);
        int gq = q0 + row, gi = ib + col;
        float dist = 3.4e38f;
        if (gq < nq && gi < it_end) {
          dist = q_sq[gq] + i_sq[gi] - 2.0f * acc[nn][r];
          dist = dist < 0.0f ? 0.0f : dist;
        }
        smem[SD + row * 129 + col] = dist;
      }
    }
    __syncthreads();

    // merge: lane t of waves 0-? — threads 0..63 each own one query row
    if (tid < KN_QB) {
      const int q = tid;
      unsigned long long* mypool = pool + q * k;
      const int lim = min(KN_IB, it_end - ib);
      for (int j = 0; j < lim; ++j) {
        float dd = smem[SD + q * 129 + j];
        if (filled < k) {
          mypool[filled] =
              ((unsigned long long)__float_as_uint(dd) << 32) | (unsigned)(ib + j);
          ++filled;
          if (filled == k) {  // establish current max
            cur_max = -1.0f;
            for (int t2 = 0; t2 < k; ++t2) {
              float v = __uint_as_float((unsigned)(mypool[t2] >> 32));
              if (v > cur_max) { cur_max = v; cur_pos = t2; }
            }
          }
        } else if (dd < cur_max) {
          mypool[cur_pos] =
              ((unsigned long long)__float_as_uint(dd) << 32) | (unsigned)(ib + j);
          cur_max = -1.0f;
          for (int t2 = 0; t2 < k; ++t2) {  // rescan for new max
            float v = __uint_as_float((unsigned)(mypool[t2] >> 32));
            if (v > cur_max) { cur_max = v; cur_pos = t2; }
          }
        }
      }
    }
    __syncthreads();
  }

  // write partials (sorted ascending by simple selection — k<=64)
  if (tid < KN_QB) {
    const int q = tid;
    if (q0 + q < nq) {
      unsigned long long* mypool = pool + q * k;
      for (int a = 0; a < k; ++a) {  // selection sort in LDS
        int best_t = a;
        unsigned long long bv = mypool[a];
        for (int b = a + 1; b < k; ++b)
          if (mypool[b] < bv) { bv = mypool[b]; best_t = b; }
        unsigned long long tmp = mypool[a];
        mypool[a] = bv;
        mypool[best_t] = tmp;
        bool invalid = (bv & 0xffffffffu) == 0xffffffffu;
        float dv = invalid ? 3.4e38f : __uint_as_float((unsigned)(bv >> 32));
        int64_t off = ((int64_t)(q0 + q) * isplit + split) * k + a;
        out_dist[off] = dv;
        out_idx[off] = invalid ? -1 : (int32_t)(bv & 0xffffffffu);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// rf_histogram: per-(node, feature, bin[, class]) histograms for random
// forest split finding (reference: cuML RF histogram build, SURVEY.md §2.3b
// tree.py:384-389). Rows pre-sorted by node (perm + seg_off, as in
// segment_sum); each (node, row-split) block accumulates its feature-chunk
// histogram in LDS (fast LDS atomics) and adds it to global once.
// Classification: C channels of class counts. Regression (n_classes==0):
// 2 channels (count, sum) — sum-of-squares cancels in the gain.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void rf_histogram_kernel(
    const uint8_t* __restrict__ Xcm,  // [d, n_phys] binned, COLUMN-major:
                                      // per-feature columns keep a segment's
                                      // row gathers inside dense cache lines
                                      // (row-major fetched ~47 64B lines per
                                      // row to use 54 sampled bytes)
    const int64_t* __restrict__ perm, // [m] VIRTUAL rows sorted by node
    const int64_t* __restrict__ seg_off,  // [B+1]
    const int32_t* __restrict__ feat_sel, // [B, mf] or nullptr (identity)
    const int32_t* __restrict__ y_cls,    // [n_phys] class ids (classification)
    const float* __restrict__ y_reg,      // [n_phys] targets (regression)
    const int32_t* __restrict__ sample,   // [vn] virtual->physical, or nullptr
    int64_t n_phys,                       // modulo map when sample==nullptr
    int d, int mf, int f0, int FC, int n_bins, int C, int split,
    float y_inv_scale,                    // regression: 1/max|y| for the
                                          // packed-u64 path; 0 = disabled
    float y_scale,                        //             max|y|
    float* __restrict__ out) {            // [B, FC, n_bins, C]
  // lhist [FC][nb][C] + an LDS copy of the block's feature-id chunk; the
  // hot loop gathers FOUR feature bytes per row before touching LDS so the
  // dependent L2/HBM gather latencies overlap (the naive one-at-a-time loop
  // ran ~12 cycles/update, far off the LDS-atomic bound)
  extern __shared__ __attribute__((aligned(16))) float lhist[];
  // per-feature COLUMN BASE OFFSETS (f * n_phys precomputed once — the
  // per-element int64 multiply cost ~3 issue slots in the hot loop);
  // region rounded up to 8B alignment (odd nfc with odd bins*classes)
  int64_t* foff_s =
      reinterpret_cast<int64_t*>(lhist + ((FC * n_bins * C + 1) & ~1));
  const int b = blockIdx.x / split;
  const int slice = blockIdx.x % split;
  const int64_t s0 = seg_off[b], e0 = seg_off[b + 1];
  const int64_t len = e0 - s0;
  const int tid = threadIdx.x;
  const int nfc = FC * n_bins * C;
  if (len == 0) {
    // split==1 gets an UNINITIALIZED out buffer (no host zero-fill): an
    // empty segment must still define its cells
    if (split == 1) {
      float* dst0 = out + (int64_t)b * nfc;
      for (int e = tid; e < nfc; e += 256) dst0[e] = 0.0f;
    }
    return;
  }
  const int64_t chunk = (len + split - 1) / split;
  const int64_t rs = s0 + slice * chunk;
  const int64_t re = min(e0, rs + chunk);
  if (rs >= re) return;
  for (int e = tid; e < nfc; e += 256) lhist[e] = 0.0f;
  for (int e = tid; e < FC; e += 256) {
    const int f = feat_sel ? feat_sel[(int64_t)b * mf + f0 + e] : (f0 + e);
    foff_s[e] = (int64_t)f * n_phys;
  }
  __syncthreads();

  const bool classif = (y_cls != nullptr);
  for (int64_t r = rs + tid; r < re; r += 256) {
    const int64_t vrow = perm[r];
    const int64_t row = sample ? (int64_t)sample[vrow]
                               : (vrow >= n_phys ? vrow % n_phys : vrow);
    const int yc = classif ? y_cls[row] : 0;
    const float yv = classif ? 1.0f : y_reg[row];
    // regression fast path: ONE u64 LDS atomic per (row, feature) — count
    // in bits 41+, biased Q14 fixed-point sum of y/max|y| below (each
    // contribution positive, so carries never cross the field boundary;
    // count <= 2^23 rows per node enforced host-side)
    const bool packed = !classif && y_inv_scale > 0.0f;
    unsigned long long yq = 0;
    if (packed) {
      const long long fix = llroundf(yv * y_inv_scale * 16384.0f) + (1ll << 15);
      yq = (1ull << 41) + (unsigned long long)fix;
    }
    int q = 0;
    for (; q + 7 < FC; q += 8) {
      int bins[8];
#pragma unroll
      for (int e = 0; e < 8; ++e)
        bins[e] = Xcm[foff_s[q + e] + row];
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        if (classif) {
          atomicAdd(&lhist[((q + e) * n_bins + bins[e]) * C + yc], 1.0f);
        } else if (packed) {
          atomicAdd(reinterpret_cast<unsigned long long*>(lhist) +
                        (q + e) * n_bins + bins[e],
                    yq);
        } else {
          float* cell = &lhist[((q + e) * n_bins + bins[e]) * 2];
          atomicAdd(cell, 1.0f);
          atomicAdd(cell + 1, yv);
        }
      }
    }
    for (; q < FC; ++q) {
      const int bin = Xcm[foff_s[q] + row];
      if (classif) {
        atomicAdd(&lhist[(q * n_bins + bin) * C + yc], 1.0f);
      } else if (packed) {
        atomicAdd(reinterpret_cast<unsigned long long*>(lhist) + q * n_bins + bin,
                  yq);
      } else {
        float* cell = &lhist[(q * n_bins + bin) * 2];
        atomicAdd(cell, 1.0f);
        atomicAdd(cell + 1, yv);
      }
    }
  }
  __syncthreads();

  float* dst = out + (int64_t)b * FC * n_bins * C;
  const bool packed_flush = (y_cls == nullptr) && y_inv_scale > 0.0f;
  if (!packed_flush) {
    if (split > 1) {
      for (int e = tid; e < nfc; e += 256)
        if (lhist[e] != 0.0f) atomicAdd(&dst[e], lhist[e]);
    } else {
      for (int e = tid; e < nfc; e += 256) dst[e] = lhist[e];
    }
  } else {
    // decode packed (count, biased Q14 sum) -> the [.., 2] float layout
    const unsigned long long* ph =
        reinterpret_cast<const unsigned long long*>(lhist);
    for (int e2 = tid; e2 < FC * n_bins; e2 += 256) {
      const unsigned long long v = ph[e2];
      const long long cnt = (long long)(v >> 41);
      const long long sq =
          (long long)(v & ((1ull << 41) - 1)) - cnt * (1ll << 15);
      const float sum = (float)sq * (y_scale / 16384.0f);
      if (split > 1) {
        if (cnt) {
          atomicAdd(&dst[e2 * 2], (float)cnt);
          atomicAdd(&dst[e2 * 2 + 1], sum);
        }
      } else {
        dst[e2 * 2] = (float)cnt;
        dst[e2 * 2 + 1] = sum;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// rf_histogram_fw: feature-wide variant — LANES run over (sorted) sampled
// features of ONE row, the loop runs over rows. The row-lane kernel above
// issued one address-divergent gather per (row, feature): each wave-load
// touched up to 64 distinct cache lines and the LSU serialized them
// (~28 cyc/update, invariant under unrolls/LDS/atomic-count experiments).
// Here a wave-load reads 32 SORTED features of one row — a short span of
// the row-major row (avg gap d/mf) — so the per-update issue cost drops by
// the lane width. Two rows per wave (lane groups of 32). Regression uses
// the packed single-u64-atomic cells.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void rf_histogram_fw_kernel(
    const uint8_t* __restrict__ Xrm,  // [n_phys, d] ROW-major
    const int64_t* __restrict__ perm,
    const int64_t* __restrict__ seg_off,  // [B+1]
    const int32_t* __restrict__ feat_sel, // [B, mf] SORTED per node, or null
    const int32_t* __restrict__ y_cls,
    const float* __restrict__ y_reg,
    const int32_t* __restrict__ sample,
    int64_t n_phys,
    int d, int mf, int f0, int FC, int n_bins, int C, int split,
    float y_inv_scale, float y_scale,
    float* __restrict__ out) {            // [B, FC, n_bins, C]
  extern __shared__ __attribute__((aligned(16))) float lhist[];
  int* fsel_s = reinterpret_cast<int*>(
      lhist + (((int64_t)FC * n_bins * C + 1) & ~1ll));
  const int b = blockIdx.x / split;
  const int slice = blockIdx.x % split;
  const int64_t s0 = seg_off[b], e0 = seg_off[b + 1];
  const int64_t len = e0 - s0;
  const int tid = threadIdx.x;
  const int nfc = FC * n_bins * C;
  const bool classif = (y_cls != nullptr);
  const bool packed = !classif && y_inv_scale > 0.0f;
  if (len == 0) {
    if (split == 1) {
      float* dst0 = out + (int64_t)b * nfc;
      for (int e = tid; e < nfc; e += 256) dst0[e] = 0.0f;
    }
    return;
  }
  const int64_t chunk = (len + split - 1) / split;
  const int64_t rs = s0 + slice * chunk;
  const int64_t re = min(e0, rs + chunk);
  if (rs >= re) return;

  for (int e = tid; e < nfc; e += 256) lhist[e] = 0.0f;
  for (int e = tid; e < FC; e += 256)
    fsel_s[e] = feat_sel ? feat_sel[(int64_t)b * mf + f0 + e] : (f0 + e);
  __syncthreads();

  // 32 rows per block iteration: 8 lane-groups x 4 INDEPENDENT row chains
  // per group — the perm->sample->row->gather dependency is ~3 serial
  // memory latencies; one chain per group left waves issue-stalled (PMC:
  // 71% SQ_WAIT_INST_ANY), four chains quadruple the loads in flight
  const int group = tid >> 5;       // 0..7
  const int fq = tid & 31;          // feature slot within the chunk
  const int f = (fq < FC) ? fsel_s[fq] : -1;
  for (int64_t base = rs; base < re; base += 32) {
    int64_t rowv[4];
    int binv[4];
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      const int64_t r = base + group + 8 * g;
      if (r < re) {
        const int64_t vrow = perm[r];
        rowv[g] = sample ? (int64_t)sample[vrow]
                         : (vrow >= n_phys ? vrow % n_phys : vrow);
      } else {
        rowv[g] = -1;
      }
    }
#pragma unroll
    for (int g = 0; g < 4; ++g)
      binv[g] = (rowv[g] >= 0 && f >= 0) ? Xrm[rowv[g] * d + f] : -1;
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      if (binv[g] < 0) continue;
      const int64_t row = rowv[g];
      const int bin = binv[g];
      if (classif) {
        atomicAdd(&lhist[(fq * n_bins + bin) * C + y_cls[row]], 1.0f);
      } else if (packed) {
        const long long fix =
            llroundf(y_reg[row] * y_inv_scale * 16384.0f) + (1ll << 15);
        atomicAdd(reinterpret_cast<unsigned long long*>(lhist) + fq * n_bins + bin,
                  (1ull << 41) + (unsigned long long)fix);
      } else {
        float* cell = &lhist[(fq * n_bins + bin) * 2];
        atomicAdd(cell, 1.0f);
        atomicAdd(cell + 1, y_reg[row]);
      }
    }
  }
  __syncthreads();

  float* dst = out + (int64_t)b * nfc;
  if (!packed) {
    if (split > 1) {
      for (int e = tid; e < nfc; e += 256)
        if (lhist[e] != 0.0f) atomicAdd(&dst[e], lhist[e]);
    } else {
      for (int e = tid; e < nfc; e += 256) dst[e] = lhist[e];
    }
  } else {
    const unsigned long long* ph =
        reinterpret_cast<const unsigned long long*>(lhist);
    for (int e2 = tid; e2 < FC * n_bins; e2 += 256) {
      const unsigned long long v = ph[e2];
      const long long cnt = (long long)(v >> 41);
      const long long sq =
          (long long)(v & ((1ull << 41) - 1)) - cnt * (1ll << 15);
      const float sum = (float)sq * (y_scale / 16384.0f);
      if (split > 1) {
        if (cnt) {
          atomicAdd(&dst[e2 * 2], (float)cnt);
          atomicAdd(&dst[e2 * 2 + 1], sum);
        }
      } else {
        dst[e2 * 2] = (float)cnt;
        dst[e2 * 2 + 1] = sum;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// rf_best_split: fused split-finding scan over rf_histogram output.
// One block per node: threads own features, scan bins accumulating left
// stats in registers, compute the gain (gini for classification; the
// cancelled-sum-of-squares form for regression), block-reduce the best
// (gain, feature, bin) and emit the winning child stats. Replaces the
// torch cumsum/where/max/gather chain (~25% of RF fit).
// C (=classes, or 2 regression channels) <= 16.
// ---------------------------------------------------------------------------

constexpr int RS_CMAX = 16;

__global__ __launch_bounds__(256) void rf_best_split_kernel(
    const float* __restrict__ H,  // [B, F, nb, C]
    int F, int nb, int C, int min_leaf, int classif,
    float* __restrict__ out_gain,   // [B]
    int32_t* __restrict__ out_feat, // [B] (chunk-local feature idx)
    int32_t* __restrict__ out_bin,  // [B]
    float* __restrict__ out_lval,   // [B, C]
    float* __restrict__ out_rval) { // [B, C]
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const float* Hb = H + (int64_t)b * F * nb * C;

  float best_gain = -1.0f;
  int best_f = -1, best_bin = -1;

  for (int f = tid; f < F; f += 256) {
    const float* Hf = Hb + (int64_t)f * nb * C;
    float total[RS_CMAX];
    for (int c = 0; c < C; ++c) total[c] = 0.0f;
    for (int bin = 0; bin < nb; ++bin)
      for (int c = 0; c < C; ++c) total[c] += Hf[bin * C + c];
    float tcnt = 0.0f, tsum = 0.0f, tsq = 0.0f;
    if (classif) {
      for (int c = 0; c < C; ++c) {
        tcnt += total[c];
        tsq += total[c] * total[c];
      }
    } else {
      tcnt = total[0];
      tsum = total[1];
    }
    if (tcnt < 2 * min_leaf) continue;

    float left[RS_CMAX];
    for (int c = 0; c < C; ++c) left[c] = 0.0f;
    float lcnt = 0.0f, lsum = 0.0f, lsq = 0.0f;
    for (int bin = 0; bin < nb - 1; ++bin) {
      if (classif) {
        for (int c = 0; c < C; ++c) {
          float v = Hf[bin * C + c];
          lsq += v * (2.0f * left[c] + v);  // maintain sum of left[c]^2
          left[c] += v;
          lcnt += v;
        }
        float rcnt = tcnt - lcnt;
        if (lcnt >= min_leaf && rcnt >= min_leaf) {
          // gini gain = g_parent - (l/t) g_l - (r/t) g_r
          float rsq = 0.0f;
          for (int c = 0; c < C; ++c) {
            float rv = total[c] - left[c];
            rsq += rv * rv;
          }
          float g_p = 1.0f - tsq / (tcnt * tcnt);
          float g_l = 1.0f - lsq / (lcnt * lcnt);
          float g_r = 1.0f - rsq / (rcnt * rcnt);
          float gain = g_p - (lcnt / tcnt) * g_l - (rcnt / tcnt) * g_r;
          if (gain > best_gain) { best_gain = gain; best_f = f; best_bin = bin; }
        }
      } else {
        lcnt += Hf[bin * C + 0];
        lsum += Hf[bin * C + 1];
        float rcnt = tcnt - lcnt, rsum = tsum - lsum;
        if (lcnt >= min_leaf && rcnt >= min_leaf) {
          float gain = (lsum * lsum / lcnt + rsum * rsum / rcnt -
                        tsum * tsum / tcnt) / tcnt;
          if (gain > best_gain) { best_gain = gain; best_f = f; best_bin = bin; }
        }
      }
    }
  }

  // block-reduce best (gain, f, bin): pack gain bits (non-negative or -1)
  __shared__ unsigned long long red[4];
  unsigned long long mine =
      ((unsigned long long)__float_as_uint(best_gain + 2.0f) << 32) |
      ((unsigned)(best_f & 0xffff) << 16) | (unsigned)(best_bin & 0xffff);
  // wave max
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    unsigned long long o = __shfl_down(mine, off, 64);
    if (o > mine) mine = o;
  }
  if ((tid & 63) == 0) red[tid >> 6] = mine;
  __syncthreads();
  if (tid == 0) {
    unsigned long long m = red[0];
    for (int w = 1; w < 4; ++w)
      if (red[w] > m) m = red[w];
    float g = __uint_as_float((unsigned)(m >> 32)) - 2.0f;
    int f = (int)((m >> 16) & 0xffff);
    int bin = (int)(m & 0xffff);
    if (f == 0xffff) { f = -1; }
    if (bin == 0xffff) { bin = -1; }
    out_gain[b] = g;
    out_feat[b] = f;
    out_bin[b] = bin;
    if (f >= 0 && bin >= 0) {
      const float* Hf = Hb + (int64_t)f * nb * C;
      for (int c = 0; c < C; ++c) {
        float l = 0.0f, t = 0.0f;
        for (int bb = 0; bb <= bin; ++bb) l += Hf[bb * C + c];
        for (int bb = 0; bb < nb; ++bb) t += Hf[bb * C + c];
        out_lval[(int64_t)b * C + c] = l;
        out_rval[(int64_t)b * C + c] = t - l;
      }
    } else {
      for (int c = 0; c < C; ++c) {
        out_lval[(int64_t)b * C + c] = 0.0f;
        out_rval[(int64_t)b * C + c] = 0.0f;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// kmeans_argmin_kn: fused epilogue over a library-GEMM dot block.
// dots is [k, n] (C @ X^T): per row i the k dot values sit at stride n, so
// the 256 threads of a block read coalesced spans per center. d2 =
// c_sq[kk] - 2*dot (x_sq[i] added once at the end). One pass over the
// [k, n] block; c_sq staged in LDS. The all-in-one MFMA assign kernel runs
// at 84 TF; hipBLASLt runs the same dot block at 141 TF, so GEMM + this
// bandwidth-bound pass wins (measured; SRML_KMEANS_VARIANT=fused keeps the
// old path).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void kmeans_argmin_kn_kernel(
    const float* __restrict__ dots,  // [k, n]
    const float* __restrict__ x_sq,  // [n]
    const float* __restrict__ c_sq,  // [k]
    int64_t n, int k,
    int32_t* __restrict__ labels,    // [n]
    float* __restrict__ min_d,       // [n] squared distance to winner
    double* __restrict__ inertia) {  // [1] accumulated
  extern __shared__ float csq_s[];
  for (int e = threadIdx.x; e < k; e += 256) csq_s[e] = c_sq[e];
  __syncthreads();
  double local_sum = 0.0;
  const int64_t stride = (int64_t)gridDim.x * 256;
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n; i += stride) {
    float best = 3.0e38f;
    int bi = 0;
    for (int kk = 0; kk < k; ++kk) {
      const float v = csq_s[kk] - 2.0f * dots[(int64_t)kk * n + i];
      if (v < best) { best = v; bi = kk; }
    }
    const float d2 = fmaxf(best + x_sq[i], 0.0f);
    labels[i] = bi;
    min_d[i] = d2;
    local_sum += (double)d2;
  }
  // block-reduce the inertia partial, one global atomic per block
  __shared__ double red_s[256];
  red_s[threadIdx.x] = local_sum;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (threadIdx.x < off) red_s[threadIdx.x] += red_s[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(inertia, red_s[0]);
}

// ---------------------------------------------------------------------------
// kmeans_argmin_nk: argmin epilogue over a ROW-major [m, k] dot block
// (dots = X_chunk @ C^T — the tall-skinny GEMM hipBLASLt runs at full rate,
// unlike the [k, n] layout whose skinny-m GEMM stalls at small k). One wave
// per row: lanes stride k contiguously (coalesced), then a shuffle
// argmin-reduce; block accumulates inertia with one atomic.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void kmeans_argmin_nk_kernel(
    const float* __restrict__ dots,  // [m, k] row-major
    const float* __restrict__ x_sq,  // [m]
    const float* __restrict__ c_sq,  // [k]
    int64_t m, int k,
    int32_t* __restrict__ labels,    // [m]
    float* __restrict__ min_d,       // [m]
    double* __restrict__ inertia) {  // [1]
  __shared__ double block_in[4];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  // grid-stride over rows: the grid is capped (<=4096 blocks) so the
  // inertia merge is one atomic per BLOCK, not per row — a block-per-row
  // launch serializes millions of f64 atomics on one word
  const int64_t wstride = (int64_t)gridDim.x * 4;
  double local = 0.0;
  for (int64_t row = (int64_t)blockIdx.x * 4 + wave; row < m; row += wstride) {
    const float* dr = dots + row * (int64_t)k;
    float best = 3.0e38f;
    int bj = 0;
    for (int j = lane; j < k; j += 64) {
      const float v = c_sq[j] - 2.0f * dr[j];
      if (v < best) { best = v; bj = j; }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      const float ob = __shfl_down(best, off, 64);
      const int oj = __shfl_down(bj, off, 64);
      if (ob < best) { best = ob; bj = oj; }
    }
    if (lane == 0) {
      const float d2 = fmaxf(best + x_sq[row], 0.0f);
      labels[row] = bj;
      min_d[row] = d2;
      local += (double)d2;
    }
  }
  if (lane == 0) block_in[wave] = local;
  __syncthreads();
  if (threadIdx.x == 0)
    atomicAdd(inertia, block_in[0] + block_in[1] + block_in[2] + block_in[3]);
}

// ---------------------------------------------------------------------------
// rf_partition: counting-sort rows by batch-local node id -> (perm, seg_off).
// Replaces the per-batch torch sort + nonzero + gather chain (radix sort was
// ~8% of RF fit kernel time; reference behavior: cuML's batched node trainer
// keeps per-node row lists, tree.py:384-389).
// Two kernels: block-privatized LDS count, then a two-pass scatter where each
// block reserves per-node ranges with ONE global atomic per (block, node)
// and places its rows via LDS-local cursors (a single global cursor per node
// would serialize ~n atomics on one word at depth 0).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void rf_node_count_kernel(
    const int64_t* __restrict__ node_of_row,  // [n]
    const int64_t* __restrict__ lut,          // [n_nodes] node -> slot | -1
    int64_t n, int B,
    int32_t* __restrict__ counts) {           // [B] pre-zeroed
  extern __shared__ int32_t lcnt0[];
  for (int e = threadIdx.x; e < B; e += 256) lcnt0[e] = 0;
  __syncthreads();
  const int64_t stride = (int64_t)gridDim.x * 256;
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n; i += stride) {
    const int64_t sl = lut[node_of_row[i]];
    if (sl >= 0) atomicAdd(&lcnt0[sl], 1);
  }
  __syncthreads();
  for (int e = threadIdx.x; e < B; e += 256)
    if (lcnt0[e]) atomicAdd(&counts[e], lcnt0[e]);
}

__global__ __launch_bounds__(256) void rf_partition_scatter_kernel(
    const int64_t* __restrict__ node_of_row,  // [n]
    const int64_t* __restrict__ lut,          // node -> slot | -1
    int64_t n, int B,
    int32_t* __restrict__ cursor,             // [B] global, init = seg_off[:B]
    int64_t* __restrict__ perm) {             // [>= total]
  extern __shared__ int32_t smem[];           // lcnt[B] + lbase[B]
  int32_t* lcnt = smem;
  int32_t* lbase = smem + B;
  for (int e = threadIdx.x; e < B; e += 256) lcnt[e] = 0;
  __syncthreads();
  const int64_t chunk = (n + gridDim.x - 1) / gridDim.x;
  const int64_t i0 = (int64_t)blockIdx.x * chunk;
  const int64_t i1 = min(n, i0 + chunk);
  for (int64_t i = i0 + threadIdx.x; i < i1; i += 256) {
    const int64_t sl = lut[node_of_row[i]];
    if (sl >= 0) atomicAdd(&lcnt[sl], 1);
  }
  __syncthreads();
  for (int e = threadIdx.x; e < B; e += 256) {
    const int c = lcnt[e];
    lbase[e] = c ? atomicAdd(&cursor[e], c) : 0;
    lcnt[e] = 0;
  }
  __syncthreads();
  for (int64_t i = i0 + threadIdx.x; i < i1; i += 256) {
    const int64_t sl = lut[node_of_row[i]];
    if (sl >= 0) {
      const int p = atomicAdd(&lcnt[sl], 1);
      perm[lbase[sl] + p] = i;
    }
  }
}

// ---------------------------------------------------------------------------
// rf_reroute: one-pass row-to-child reassignment after a split batch.
// Replaces the torch lut-gather + nonzero + feature-gather + where chain
// (the byte index_elementwise gathers were ~17% of RFC fit kernel time).
// Training rule: bin <= split_bin goes LEFT (== x < edges[split_bin]).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void rf_reroute_kernel(
    int64_t* __restrict__ node_of_row,        // [vn] in/out (virtual rows)
    const int64_t* __restrict__ lut2,         // node -> split slot | -1
    const int32_t* __restrict__ sfeat,        // [ns]
    const int32_t* __restrict__ sbin,         // [ns]
    const int64_t* __restrict__ lchild,       // [ns]
    const int64_t* __restrict__ rchild,       // [ns]
    const uint8_t* __restrict__ Xcm,          // [d, n_phys] column-major
    const int32_t* __restrict__ sample,       // [vn] or nullptr
    int64_t n_phys,
    int64_t vn, int64_t d) {
  const int64_t stride = (int64_t)gridDim.x * 256;
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < vn; i += stride) {
    const int64_t sl = lut2[node_of_row[i]];
    if (sl < 0) continue;
    const int64_t row = sample ? (int64_t)sample[i]
                               : (i >= n_phys ? i % n_phys : i);
    const uint8_t bin = Xcm[(int64_t)sfeat[sl] * n_phys + row];
    node_of_row[i] = (bin <= (uint8_t)sbin[sl]) ? lchild[sl] : rchild[sl];
  }
}

// ---------------------------------------------------------------------------
// knn_merge_topk: running per-query top-k selection fused with the
// ||q-i||^2 = q_sq + i_sq - 2*dot expansion, consuming a library-GEMM dot
// block ONCE (the torch path materializes d2 and runs multi-pass topk +
// cat/gather merges per item chunk). One wave per query row; each lane owns
// one of 64 candidate slots; tau = current worst kept distance, maintained
// by wave max-reduce. Insertions serialize per wave but decay to rare as tau
// shrinks (expected k*ln(c/k) per row on random-order data).
// Reference: NearestNeighborsMG tiled distance + topk merge (knn.py:763-774).
// ---------------------------------------------------------------------------

__device__ inline float wave_max_f32(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

__global__ __launch_bounds__(256) void knn_merge_topk_kernel(
    const float* __restrict__ G,     // [nq, c] dot products Q @ I_chunk^T
    const float* __restrict__ q_sq,  // [nq]
    const float* __restrict__ i_sq,  // [c]
    int64_t nq, int64_t c, int64_t base,
    float* __restrict__ best_d,      // [nq, 64] slots (+inf live, -inf dead)
    int64_t* __restrict__ best_i) {  // [nq, 64]
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int64_t row = (int64_t)blockIdx.x * 4 + wave;
  if (row >= nq) return;

  float cur_d = best_d[row * 64 + lane];
  int64_t cur_i = best_i[row * 64 + lane];
  float tau = wave_max_f32(cur_d);

  const float* g = G + row * c;
  const float qs = q_sq[row];

  // float4 loads are legal only when every row base is 16B-aligned, i.e.
  // c % 4 == 0 (g = G + row*c); otherwise scalar loads
  const bool vec4 = (c & 3) == 0;
  // UNIFORM trip count: every lane of the wave must reach every ballot /
  // shuffle below even when its own j0 has run past c (a per-lane `j0 < c`
  // loop bound left part of the wave outside the evict sequence whenever
  // c % 256 != 0, corrupting tau)
  const int64_t n_iter = (c + 255) / 256;
  for (int64_t it = 0; it < n_iter; ++it) {
    const int64_t j0 = it * 256 + (int64_t)lane * 4;
    float dot4[4], isq4[4];
    if (vec4 && j0 + 3 < c) {
      const float4 gd = *reinterpret_cast<const float4*>(g + j0);
      const float4 gi = *reinterpret_cast<const float4*>(i_sq + j0);
      dot4[0] = gd.x; dot4[1] = gd.y; dot4[2] = gd.z; dot4[3] = gd.w;
      isq4[0] = gi.x; isq4[1] = gi.y; isq4[2] = gi.z; isq4[3] = gi.w;
    } else {
      for (int e = 0; e < 4; ++e) {
        const bool ok = j0 + e < c;
        dot4[e] = ok ? g[j0 + e] : 0.0f;
        isq4[e] = ok ? i_sq[j0 + e] : 3.0e38f;  // pushes d2 above any tau
      }
    }
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      const float d2 = qs + isq4[e] - 2.0f * dot4[e];
      unsigned long long m = __ballot(d2 < tau && j0 + e < c);
      while (m) {
        const int src = __ffsll((unsigned long long)m) - 1;
        m &= m - 1;
        const float v = __shfl(d2, src, 64);
        if (v >= tau) continue;  // tau shrank since the ballot
        const int col = __shfl((int)(j0 + e), src, 64);
        // evict the current worst slot
        const unsigned long long owners = __ballot(cur_d == tau);
        const int owner = __ffsll((unsigned long long)owners) - 1;
        if (lane == owner) { cur_d = v; cur_i = base + col; }
        tau = wave_max_f32(cur_d);
      }
    }
  }

  best_d[row * 64 + lane] = cur_d;
  best_i[row * 64 + lane] = cur_i;
}

// ---------------------------------------------------------------------------
// gather_dists: d2[i][j] = ||A[i] - B[cand[i][j]]||^2 — the nn-descent /
// beam-search hot op (per-row private candidate lists make this a gathered
// row-vs-rows reduction, not a GEMM). One wave per output row: A[i] is
// staged in registers (d/64 floats per lane), each candidate's B row is read
// coalesced by the wave and reduced with a 6-step butterfly. The torch path
// materializes [r, m, d] twice; this reads each gathered row once.
// Reference: cuVS nn_descent / cagra build (umap.py:359-378).
// ---------------------------------------------------------------------------

constexpr int GD_DMAX = 2048;  // register-staged A row: d/64 <= 32 VGPRs

__global__ __launch_bounds__(256) void gather_dists_kernel(
    const float* __restrict__ A,     // [r, d]
    const float* __restrict__ B,     // [n, d]
    const int64_t* __restrict__ cand,  // [r, m]
    int64_t r, int64_t m, int d,
    float* __restrict__ out) {       // [r, m]
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int64_t row = (int64_t)blockIdx.x * 4 + wave;
  if (row >= r) return;

  const int nseg = (d + 63) >> 6;  // elements per lane
  float a_reg[GD_DMAX / 64];
  const float* arow = A + row * d;
#pragma unroll 4
  for (int s = 0; s < nseg; ++s) {
    const int idx = s * 64 + lane;
    a_reg[s] = (idx < d) ? arow[idx] : 0.0f;
  }

  const int64_t* crow = cand + row * m;
  for (int64_t j = 0; j < m; ++j) {
    const float* brow = B + crow[j] * d;
    float acc = 0.0f;
#pragma unroll 4
    for (int s = 0; s < nseg; ++s) {
      const int idx = s * 64 + lane;
      const float bv = (idx < d) ? brow[idx] : 0.0f;
      const float diff = a_reg[s] - bv;
      acc = fmaf(diff, diff, acc);
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) acc += __shfl_xor(acc, off, 64);
    if (lane == 0) out[row * m + j] = acc;
  }
}

// ---------------------------------------------------------------------------
// CSR GLM kernels (the reference's LogisticRegressionMG sparse path,
// classification.py:960-966, runs cuML SpMM; torch/rocSPARSE here spent its
// time in f64 index_add column stats, a 400M-pair radix-sort transpose and
// a generic csrmm — these three passes replace all of it):
//  - csr_fwd:        scores[n,C] = A @ WT   (thread per row, ~nnz/row loop)
//  - csr_grad:       grad[C,d]  += val * resid[row,C] scattered by column
//                    (2048-way atomic fan-out pipelines fine; removes the
//                    pre-transposed CSR entirely)
//  - csr_col_moments: per-column sum / sum-of-squares in one pass
// ---------------------------------------------------------------------------

template <typename IdxT>
__global__ __launch_bounds__(256) void csr_fwd_kernel(
    const IdxT* __restrict__ indptr, const IdxT* __restrict__ indices,
    const float* __restrict__ vals, const float* __restrict__ WT,  // [d, C]
    int64_t n, int C, float* __restrict__ scores) {                // [n, C]
  const int64_t stride = (int64_t)gridDim.x * 256;
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n; i += stride) {
    float acc[32];
    for (int c = 0; c < C; ++c) acc[c] = 0.0f;
    const int64_t e0 = (int64_t)indptr[i], e1 = (int64_t)indptr[i + 1];
    for (int64_t j = e0; j < e1; ++j) {
      const float v = vals[j];
      const float* wrow = WT + (int64_t)indices[j] * C;
      for (int c = 0; c < C; ++c) acc[c] = fmaf(v, wrow[c], acc[c]);
    }
    for (int c = 0; c < C; ++c) scores[i * C + c] = acc[c];
  }
}

template <typename IdxT>
__global__ __launch_bounds__(256) void csr_grad_kernel(
    const IdxT* __restrict__ indptr, const IdxT* __restrict__ indices,
    const float* __restrict__ vals, const float* __restrict__ resid,  // [n, C]
    int64_t n, int64_t d, int C, int use_lds,
    float* __restrict__ grad) {          // [C, d]
  // grad is tiny ([C, d] ~ KBs): accumulate per BLOCK in LDS and flush once
  // — 400M chip-wide global atomics onto 2048 words ran at ~8 G/s (48 ms
  // per iteration); LDS-privatized they cost ~5 ms
  extern __shared__ float g_s[];
  const int64_t cd = (int64_t)C * d;
  if (use_lds) {
    for (int64_t e = threadIdx.x; e < cd; e += 256) g_s[e] = 0.0f;
    __syncthreads();
  }
  float* acc = use_lds ? g_s : grad;
  const int64_t stride = (int64_t)gridDim.x * 256;
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n; i += stride) {
    float r[32];
    for (int c = 0; c < C; ++c) r[c] = resid[i * C + c];
    const int64_t e0 = (int64_t)indptr[i], e1 = (int64_t)indptr[i + 1];
    for (int64_t j = e0; j < e1; ++j) {
      const float v = vals[j];
      const int64_t col = (int64_t)indices[j];
      for (int c = 0; c < C; ++c) atomicAdd(&acc[(int64_t)c * d + col], v * r[c]);
    }
  }
  if (use_lds) {
    __syncthreads();
    for (int64_t e = threadIdx.x; e < cd; e += 256)
      if (g_s[e] != 0.0f) atomicAdd(&grad[e], g_s[e]);
  }
}

template <typename IdxT>
__global__ __launch_bounds__(256) void csr_col_moments_kernel(
    const IdxT* __restrict__ indices, const float* __restrict__ vals,
    int64_t nnz, int64_t d, double* __restrict__ out) {  // [2, d] sum, sumsq
  extern __shared__ double cm_s[];  // [2, d] per-block
  for (int64_t e = threadIdx.x; e < 2 * d; e += 256) cm_s[e] = 0.0;
  __syncthreads();
  const int64_t stride = (int64_t)gridDim.x * 256;
  for (int64_t j = (int64_t)blockIdx.x * 256 + threadIdx.x; j < nnz; j += stride) {
    const double v = (double)vals[j];
    const int64_t c = (int64_t)indices[j];
    atomicAdd(&cm_s[c], v);
    atomicAdd(&cm_s[d + c], v * v);
  }
  __syncthreads();
  for (int64_t e = threadIdx.x; e < 2 * d; e += 256)
    if (cm_s[e] != 0.0) atomicAdd(&out[e], cm_s[e]);
}

// ---------------------------------------------------------------------------
// fil_predict: forest inference (the reference's FIL predict kernel,
// tree.py:709-721 / cuML FIL). All trees live flattened in one node arena
// (per-tree root offsets); one THREAD walks every tree for its row,
// accumulating leaf values (class-count vote or regression mean). Routing
// rule is the training one: x < threshold goes LEFT.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void fil_predict_kernel(
    const float* __restrict__ X,        // [n, d]
    const int32_t* __restrict__ feat,   // arena [total_nodes]
    const float* __restrict__ thr,      // arena
    const int32_t* __restrict__ left,   // arena (tree-local ids)
    const int32_t* __restrict__ right,  // arena
    const float* __restrict__ value,    // arena [total_nodes, vw]
    const int64_t* __restrict__ roots,  // [T] arena offset of each tree root
    int64_t n, int d, int T, int vw, int classif,
    float* __restrict__ out) {          // [n, vw] summed votes / means
  const int64_t stride = (int64_t)gridDim.x * 256;
  float acc[16];  // vw <= 16 (host falls back to torch past that)
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n; i += stride) {
    const float* xrow = X + i * d;
    for (int c = 0; c < vw; ++c) acc[c] = 0.0f;
    for (int t = 0; t < T; ++t) {
      const int64_t base = roots[t];
      int node = 0;
      int f = feat[base + node];
      while (f >= 0) {
        node = (xrow[f] < thr[base + node]) ? left[base + node]
                                            : right[base + node];
        f = feat[base + node];
      }
      const float* v = value + (base + node) * vw;
      if (classif) {
        // normalized class-count vote (matches the torch traversal)
        float s = 0.0f;
        for (int c = 0; c < vw; ++c) s += v[c];
        s = fmaxf(s, 1e-12f);
        for (int c = 0; c < vw; ++c) acc[c] += v[c] / s;
      } else {
        acc[0] += v[0];  // (mean, count) leaves: mean is v[0]
      }
    }
    if (classif) {
      for (int c = 0; c < vw; ++c) out[i * vw + c] = acc[c];
    } else {
      out[i * vw + 0] = acc[0];
    }
  }
}

// ---------------------------------------------------------------------------
// softmax_residual_loss: per-row softmax (C>1) or sigmoid (C==1) residual
// and summed log-loss. resid = softmax(scores) - onehot(y) (or p - y).
// One wave per row chunk; memory-bound, fused to one pass.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void softmax_residual_kernel(
    const float* __restrict__ scores,  // [n, C]
    const int64_t* __restrict__ y,     // [n]
    int64_t n, int C,
    float* __restrict__ resid,         // [n, C]
    float* __restrict__ loss) {        // [1]
  const int64_t row0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float local_loss = 0.0f;
  for (int64_t i = row0; i < n; i += stride) {
    const float* s = scores + i * C;
    float* r = resid + i * C;
    const int yi = (int)y[i];
    if (C == 1) {
      float z = s[0];
      float t = yi ? 1.0f : -1.0f;
      // log(1+exp(-t z)) stable
      float m = -t * z;
      local_loss += (m > 0 ? m : 0.0f) + __logf(1.0f + __expf(-fabsf(m)));
      float p = 1.0f / (1.0f + __expf(-z));
      r[0] = p - (float)yi;
    } else {
      float mx = s[0];
      for (int c = 1; c < C; ++c) mx = fmaxf(mx, s[c]);
      float denom = 0.0f;
      for (int c = 0; c < C; ++c) denom += __expf(s[c] - mx);
      float logd = __logf(denom);
      local_loss += -(s[yi] - mx - logd);
      for (int c = 0; c < C; ++c) {
        float p = __expf(s[c] - mx) / denom;
        r[c] = p - (c == yi ? 1.0f : 0.0f);
      }
    }
  }
  // block-reduce the loss, one atomic per wave
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    local_loss += __shfl_down(local_loss, off, 64);
  if ((threadIdx.x & 63) == 0) atomicAdd(loss, local_loss);
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

static inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}


torch::Tensor dbscan_sweep(torch::Tensor X, torch::Tensor x_sq, int64_t row0,
                           int64_t n_rows, double eps2, int64_t mode,
                           torch::Tensor core, torch::Tensor labels,
                           torch::Tensor tile_idx, torch::Tensor tile_off) {
  TORCH_CHECK(X.is_cuda() && x_sq.is_cuda(), "device tensors required");
  TORCH_CHECK(X.dtype() == torch::kFloat32 && X.is_contiguous());
  const int64_t n = X.size(0);
  const int d = (int)X.size(1);
  auto out = torch::empty({n_rows}, X.options().dtype(torch::kInt32));
  if (n_rows == 0) return out;
  TORCH_CHECK(row0 >= 0 && row0 + n_rows <= n, "row slice out of range");
  if (mode == 1) {
    TORCH_CHECK(core.dtype() == torch::kUInt8 && core.is_contiguous());
    TORCH_CHECK(labels.dtype() == torch::kInt32 && labels.is_contiguous());
  }
  const int grid = (int)((n_rows + KM_BM - 1) / KM_BM);
  const bool pruned = tile_off.numel() > 0;
  if (pruned) {
    TORCH_CHECK(tile_idx.dtype() == torch::kInt32 && tile_idx.is_contiguous());
    TORCH_CHECK(tile_off.dtype() == torch::kInt32 && tile_off.is_contiguous());
    TORCH_CHECK(tile_off.numel() == grid + 1, "tile_off must have one entry per row block + 1");
  }
  hipLaunchKernelGGL(dbscan_sweep_kernel, dim3(grid), dim3(256), 0, cur_stream(),
                     X.data_ptr<float>(), x_sq.data_ptr<float>(), (int)n, d,
                     (int)row0, (int)n_rows, (float)eps2, (int)mode,
                     mode == 1 ? core.data_ptr<uint8_t>() : nullptr,
                     mode == 1 ? labels.data_ptr<int32_t>() : nullptr,
                     pruned ? tile_idx.data_ptr<int32_t>() : nullptr,
                     pruned ? tile_off.data_ptr<int32_t>() : nullptr,
                     out.data_ptr<int32_t>());
  return out;
}


torch::Tensor umap_sgd(torch::Tensor emb, torch::Tensor tail_emb,
                       torch::Tensor heads, torch::Tensor tails,
                       torch::Tensor eps, int64_t n_epochs, double a, double b,
                       double lr, double repulsion, int64_t neg_rate,
                       bool move_tail, int64_t seed) {
  TORCH_CHECK(emb.is_cuda() && emb.dtype() == torch::kFloat32 && emb.is_contiguous());
  TORCH_CHECK(heads.dtype() == torch::kInt32 && tails.dtype() == torch::kInt32);
  TORCH_CHECK(eps.dtype() == torch::kFloat32);
  const int m = (int)heads.size(0);
  const int dim = (int)emb.size(1);
  const int n_vertices = (int)tail_emb.size(0);
  TORCH_CHECK(dim >= 1 && dim <= 4, "umap_sgd supports dim 1..4");
  auto next_due = eps.clone();
  const int grid = (m + 255) / 256;
  for (int epoch = 1; epoch <= (int)n_epochs; ++epoch) {
    const float alpha = (float)(lr * (1.0 - (double)epoch / (double)n_epochs));
#define UMAP_LAUNCH(D)                                                         \
    hipLaunchKernelGGL(umap_sgd_epoch_kernel<D>, dim3(grid), dim3(256), 0,     \
                       cur_stream(), emb.data_ptr<float>(),                    \
                       tail_emb.data_ptr<float>(), heads.data_ptr<int32_t>(),  \
                       tails.data_ptr<int32_t>(), eps.data_ptr<float>(),       \
                       next_due.data_ptr<float>(), m, n_vertices, (float)a,    \
                       (float)b, alpha, (float)repulsion, (int)neg_rate,       \
                       move_tail ? 1 : 0, epoch, (unsigned)(seed & 0xffffffff))
    switch (dim) {
      case 1: UMAP_LAUNCH(1); break;
      case 2: UMAP_LAUNCH(2); break;
      case 3: UMAP_LAUNCH(3); break;
      default: UMAP_LAUNCH(4); break;
    }
#undef UMAP_LAUNCH
  }
  return emb;
}

std::vector<torch::Tensor> kmeans_assign(torch::Tensor X, torch::Tensor C,
                                         torch::Tensor x_sq) {
  TORCH_CHECK(X.is_cuda() && C.is_cuda() && x_sq.is_cuda(), "device tensors required");
  TORCH_CHECK(X.dtype() == torch::kFloat32 && C.dtype() == torch::kFloat32);
  TORCH_CHECK(X.is_contiguous() && C.is_contiguous());
  const int64_t n = X.size(0);
  const int d = (int)X.size(1);
  const int k_real = (int)C.size(0);
  auto c_sq = (C * C).sum(1);
  // Pad centers to a KM_BN multiple: a partial center tile routes BOTH the
  // X and C stages of that tile through the guarded per-element load path
  // (the de-pipelining trap in profiles/README.md) — at k=200 that is half
  // of all tiles and cost ~2x (209 ms vs ~100 ms on 100M x 128). Pad rows
  // carry c_sq = FLT_MAX so they can never win the argmin.
  const int k = (int)((k_real + KM_BN - 1) / KM_BN * KM_BN);
  if (k != k_real) {
    auto Cp = torch::zeros({(int64_t)k, (int64_t)d}, C.options());
    Cp.narrow(0, 0, k_real).copy_(C);
    auto cp = torch::full({(int64_t)k}, 3.0e38f, c_sq.options());
    cp.narrow(0, 0, k_real).copy_(c_sq);
    C = Cp.contiguous();
    c_sq = cp.contiguous();
  }
  auto labels = torch::empty({n}, X.options().dtype(torch::kInt32));
  auto min_dists = torch::empty({n}, X.options());
  auto inertia = torch::zeros({1}, X.options().dtype(torch::kFloat64));
  const int grid = (int)((n + KM_BM - 1) / KM_BM);
  if (n > 0) {
    // A/B-measured on MI355X (1M x 3000, k=1000): register-staged BK=64
    // write-after-barrier 71.1 ms; glds 2-buffer 73.8 ms; glds 3-buffer
    // 1-block/CU span 111 ms (single wave/SIMD starves the MFMA pipe).
    // Register staging is the default; glds variants stay selectable.
    static const char* v = getenv("SRML_KMEANS_VARIANT");
    if (v && v[0] == 'b' && d % 4 == 0) {  // big-tile 256x256 glds variant
      const int grid2 = (int)((n + KG2_BM - 1) / KG2_BM);
      hipLaunchKernelGGL(kmeans_assign_256_kernel, dim3(grid2), dim3(512), 0,
                         cur_stream(), X.data_ptr<float>(), C.data_ptr<float>(),
                         x_sq.data_ptr<float>(), c_sq.data_ptr<float>(), (int)n,
                         d, k, labels.data_ptr<int32_t>(),
                         min_dists.data_ptr<float>(),
                         inertia.data_ptr<double>());
    } else if (d % 4 == 0 && v && v[0] == 's') {
      const int grid_s = (int)((n + KS_BM - 1) / KS_BM);
      hipLaunchKernelGGL(kmeans_assign_sb3_kernel, dim3(grid_s), dim3(256), 0,
                         cur_stream(), X.data_ptr<float>(), C.data_ptr<float>(),
                         x_sq.data_ptr<float>(), c_sq.data_ptr<float>(), (int)n, d, k,
                         labels.data_ptr<int32_t>(), min_dists.data_ptr<float>(),
                         inertia.data_ptr<double>());
    } else if (d % 4 == 0 && v && v[0] == '3')
      hipLaunchKernelGGL(kmeans_assign_glds3_kernel, dim3(grid), dim3(256), 0,
                         cur_stream(), X.data_ptr<float>(), C.data_ptr<float>(),
                         x_sq.data_ptr<float>(), c_sq.data_ptr<float>(), (int)n, d, k,
                         labels.data_ptr<int32_t>(), min_dists.data_ptr<float>(),
                         inertia.data_ptr<double>());
    else if (d % 4 == 0 && v && v[0] == '2')
      hipLaunchKernelGGL(kmeans_assign_glds_kernel, dim3(grid), dim3(256), 0,
                         cur_stream(), X.data_ptr<float>(), C.data_ptr<float>(),
                         x_sq.data_ptr<float>(), c_sq.data_ptr<float>(), (int)n, d, k,
                         labels.data_ptr<int32_t>(), min_dists.data_ptr<float>(),
                         inertia.data_ptr<double>());
    else
      hipLaunchKernelGGL(kmeans_assign_kernel, dim3(grid), dim3(256), 0, cur_stream(),
                         X.data_ptr<float>(), C.data_ptr<float>(), x_sq.data_ptr<float>(),
                         c_sq.data_ptr<float>(), (int)n, d, k,
                         labels.data_ptr<int32_t>(), min_dists.data_ptr<float>(),
                         inertia.data_ptr<double>());
  }
  return {labels, min_dists, inertia};
}

std::vector<torch::Tensor> label_accumulate(torch::Tensor X, torch::Tensor labels, int64_t k) {
  TORCH_CHECK(X.is_cuda() && labels.is_cuda());
  const int64_t n = X.size(0);
  const int d = (int)X.size(1);
  auto sums = torch::zeros({k, d}, X.options());
  auto counts = torch::zeros({k}, X.options());
  if (n == 0) return {sums, counts};
  const int64_t kd4 = k * (int64_t)d * 4;
  if (kd4 + k * 4 <= 120 * 1024 && labels.dtype() == torch::kInt32) {
    const unsigned grid = (unsigned)std::min<int64_t>(512, (n + 511) / 512 + 1);
    const int64_t rows_per_block = (n + grid - 1) / grid;
    const size_t lds = (size_t)kd4 + (size_t)k * 4;
    hipLaunchKernelGGL(label_accumulate_lds_kernel, dim3(grid), dim3(512), lds,
                       cur_stream(), X.data_ptr<float>(),
                       labels.data_ptr<int32_t>(), n, d, (int)k,
                       rows_per_block, sums.data_ptr<float>(),
                       counts.data_ptr<float>());
    return {sums, counts};
  }
  {
    auto sorted = labels.to(torch::kInt64).sort();
    auto perm = std::get<1>(sorted).contiguous();
    auto sl = std::get<0>(sorted).contiguous();
    auto bounds = torch::arange(k + 1, labels.options().dtype(torch::kInt64));
    auto seg_off = torch::searchsorted(sl, bounds).contiguous();
    hipLaunchKernelGGL(segment_sum_kernel, dim3((unsigned)(k * LA_SPLIT)), dim3(256), 0,
                       cur_stream(), X.data_ptr<float>(), perm.data_ptr<int64_t>(),
                       seg_off.data_ptr<int64_t>(), d, sums.data_ptr<float>(),
                       counts.data_ptr<float>());
  }
  return {sums, counts};
}

torch::Tensor gram_f32(torch::Tensor A) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kFloat32 && A.is_contiguous());
  const int64_t n = A.size(0);
  const int d = (int)A.size(1);
  auto out = torch::zeros({d, d}, A.options());
  if (n > 0 && d > 0) {
    const int nb = (d + GR_BN - 1) / GR_BN;
    const int tri = nb * (nb + 1) / 2;
    int split = std::max(1, std::min(64, 768 / tri));
    dim3 grid((unsigned)(tri * split));
    hipLaunchKernelGGL(gram_kernel, grid, dim3(256), 0, cur_stream(),
                       A.data_ptr<float>(), n, d, nb, split, out.data_ptr<float>());
    int mgrid = (int)std::min<int64_t>(2048, ((int64_t)d * d + 255) / 256);
    hipLaunchKernelGGL(gram_mirror_kernel, dim3(mgrid), dim3(256), 0, cur_stream(),
                       out.data_ptr<float>(), d);
  }
  return out;
}

std::vector<torch::Tensor> knn_select(torch::Tensor Q, torch::Tensor I, int64_t k) {
  TORCH_CHECK(Q.is_cuda() && I.is_cuda());
  TORCH_CHECK(Q.dtype() == torch::kFloat32 && I.dtype() == torch::kFloat32);
  TORCH_CHECK(Q.is_contiguous() && I.is_contiguous());
  TORCH_CHECK(k >= 1 && k <= KN_KMAX, "knn_select supports k in [1, 64]");
  const int nq = (int)Q.size(0);
  const int ni = (int)I.size(0);
  const int d = (int)Q.size(1);
  auto q_sq = (Q * Q).sum(1);
  auto i_sq = (I * I).sum(1);
  const int qblocks = (nq + KN_QB - 1) / KN_QB;
  int isplit = std::max(1, std::min((ni + KN_IB - 1) / KN_IB, 512 / std::max(1, qblocks)));
  auto dist = torch::empty({nq, (int64_t)isplit, k}, Q.options());
  auto idx = torch::empty({nq, (int64_t)isplit, k}, Q.options().dtype(torch::kInt32));
  if (nq > 0 && ni > 0)
    hipLaunchKernelGGL(knn_select_kernel, dim3((unsigned)(qblocks * isplit)), dim3(256),
                       0, cur_stream(), Q.data_ptr<float>(), I.data_ptr<float>(),
                       q_sq.data_ptr<float>(), i_sq.data_ptr<float>(), nq, ni, d,
                       (int)k, isplit, dist.data_ptr<float>(), idx.data_ptr<int32_t>());
  // merge partial top-k across item splits
  auto dflat = dist.view({nq, (int64_t)isplit * k});
  auto iflat = idx.view({nq, (int64_t)isplit * k});
  int64_t kk = std::min<int64_t>(k, (int64_t)isplit * k);
  auto top = dflat.topk(kk, /*dim=*/1, /*largest=*/false);
  auto vals = std::get<0>(top);
  auto order = std::get<1>(top);
  auto ids = iflat.gather(1, order);
  return {vals, ids.to(torch::kInt64)};
}

std::vector<torch::Tensor> softmax_residual_loss(torch::Tensor scores, torch::Tensor y) {
  TORCH_CHECK(scores.is_cuda() && y.is_cuda());
  TORCH_CHECK(scores.dtype() == torch::kFloat32);
  TORCH_CHECK(y.dtype() == torch::kInt64);
  const int64_t n = scores.size(0);
  const int C = scores.dim() > 1 ? (int)scores.size(1) : 1;
  auto resid = torch::empty_like(scores);
  auto loss = torch::zeros({1}, scores.options());
  if (n > 0) {
    int grid = (int)std::min<int64_t>(2048, (n + 255) / 256);
    hipLaunchKernelGGL(softmax_residual_kernel, dim3(grid), dim3(256), 0, cur_stream(),
                       scores.data_ptr<float>(), y.data_ptr<int64_t>(), n, C,
                       resid.data_ptr<float>(), loss.data_ptr<float>());
  }
  return {resid, loss.squeeze(0)};
}

torch::Tensor rf_histogram(torch::Tensor Xb, torch::Tensor perm, torch::Tensor seg_off,
                           torch::Tensor feat_sel, torch::Tensor y, int64_t f0,
                           int64_t FC, int64_t n_bins, int64_t n_classes,
                           torch::Tensor sample, double y_max) {
  // Xb here is the COLUMN-major binned matrix [d, n_phys]
  TORCH_CHECK(Xb.is_cuda() && Xb.dtype() == torch::kUInt8 && Xb.is_contiguous());
  TORCH_CHECK(perm.dtype() == torch::kInt64 && seg_off.dtype() == torch::kInt64);
  const int d = (int)Xb.size(0);
  const int64_t n_phys = Xb.size(1);
  const int B = (int)seg_off.size(0) - 1;
  const bool classif = n_classes > 0;
  const int C = classif ? (int)n_classes : 2;
  const int mf = feat_sel.numel() > 0 ? (int)feat_sel.size(1) : 0;
  const bool has_sample = sample.numel() > 0;
  if (has_sample) TORCH_CHECK(sample.dtype() == torch::kInt32 && sample.is_contiguous());
  // regression packed-u64 path: caller supplies max|y| (computed once per
  // forest fit — a per-call .item() would sync every chunk) and the
  // per-node row-count cap gates it
  float y_scale = 0.0f, y_inv_scale = 0.0f;
  if (!classif && y_max > 0.0) {
    const int64_t vn = has_sample ? sample.numel() : perm.numel();
    if (vn < (1ll << 23)) {
      y_scale = (float)y_max;
      y_inv_scale = 1.0f / y_scale;
    }
  }
  const size_t lds = ((size_t)FC * n_bins * C + 1 & ~1ull) * 4 + (size_t)FC * 8;  // hist + col offsets
  TORCH_CHECK(lds <= 160 * 1024, "feature chunk too large for LDS");
  int split = std::max(1, (int)(1024 / std::max(1, B)));
  // split==1 writes every cell exactly once -> skip the zero-fill kernel
  auto out = split > 1
      ? torch::zeros({(int64_t)B, FC, n_bins, (int64_t)C}, Xb.options().dtype(torch::kFloat32))
      : torch::empty({(int64_t)B, FC, n_bins, (int64_t)C}, Xb.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(rf_histogram_kernel, dim3((unsigned)(B * split)), dim3(256), lds,
                     cur_stream(), Xb.data_ptr<uint8_t>(), perm.data_ptr<int64_t>(),
                     seg_off.data_ptr<int64_t>(),
                     mf > 0 ? feat_sel.data_ptr<int32_t>() : nullptr,
                     classif ? y.data_ptr<int32_t>() : nullptr,
                     classif ? nullptr : y.data_ptr<float>(),
                     has_sample ? sample.data_ptr<int32_t>() : nullptr, n_phys,
                     d, mf, (int)f0, (int)FC, (int)n_bins, C, split,
                     y_inv_scale, y_scale,
                     out.data_ptr<float>());
  return out;
}

torch::Tensor rf_histogram_fw(torch::Tensor Xrm, torch::Tensor perm, torch::Tensor seg_off,
                              torch::Tensor feat_sel, torch::Tensor y, int64_t f0,
                              int64_t FC, int64_t n_bins, int64_t n_classes,
                              torch::Tensor sample, double y_max) {
  // Xrm is ROW-major [n_phys, d]
  TORCH_CHECK(Xrm.is_cuda() && Xrm.dtype() == torch::kUInt8 && Xrm.is_contiguous());
  TORCH_CHECK(perm.dtype() == torch::kInt64 && seg_off.dtype() == torch::kInt64);
  TORCH_CHECK(FC <= 32, "rf_histogram_fw: FC <= 32 (one lane group)");
  const int d = (int)Xrm.size(1);
  const int64_t n_phys = Xrm.size(0);
  const int B = (int)seg_off.size(0) - 1;
  const bool classif = n_classes > 0;
  const int C = classif ? (int)n_classes : 2;
  const int mf = feat_sel.numel() > 0 ? (int)feat_sel.size(1) : 0;
  const bool has_sample = sample.numel() > 0;
  float y_scale = 0.0f, y_inv_scale = 0.0f;
  if (!classif && y_max > 0.0) {
    const int64_t vn = has_sample ? sample.numel() : perm.numel();
    if (vn < (1ll << 23)) {
      y_scale = (float)y_max;
      y_inv_scale = 1.0f / y_scale;
    }
  }
  const size_t lds = (((size_t)FC * n_bins * C + 1) & ~1ull) * 4 + (size_t)FC * 4;
  TORCH_CHECK(lds <= 160 * 1024, "feature chunk too large for LDS");
  int split = std::max(1, (int)(1024 / std::max(1, B)));
  auto out = split > 1
      ? torch::zeros({(int64_t)B, FC, n_bins, (int64_t)C}, Xrm.options().dtype(torch::kFloat32))
      : torch::empty({(int64_t)B, FC, n_bins, (int64_t)C}, Xrm.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(rf_histogram_fw_kernel, dim3((unsigned)(B * split)), dim3(256), lds,
                     cur_stream(), Xrm.data_ptr<uint8_t>(), perm.data_ptr<int64_t>(),
                     seg_off.data_ptr<int64_t>(),
                     mf > 0 ? feat_sel.data_ptr<int32_t>() : nullptr,
                     classif ? y.data_ptr<int32_t>() : nullptr,
                     classif ? nullptr : y.data_ptr<float>(),
                     has_sample ? sample.data_ptr<int32_t>() : nullptr, n_phys,
                     d, mf, (int)f0, (int)FC, (int)n_bins, C, split,
                     y_inv_scale, y_scale,
                     out.data_ptr<float>());
  return out;
}

std::vector<torch::Tensor> rf_best_split(torch::Tensor H, int64_t min_leaf,
                                          bool classif) {
  TORCH_CHECK(H.is_cuda() && H.dtype() == torch::kFloat32 && H.is_contiguous());
  const int B = (int)H.size(0);
  const int F = (int)H.size(1);
  const int nb = (int)H.size(2);
  const int C = (int)H.size(3);
  TORCH_CHECK(C <= RS_CMAX, "rf_best_split supports C <= 16");
  TORCH_CHECK(F < 0xffff && nb < 0xffff);
  auto gain = torch::empty({B}, H.options());
  auto feat = torch::empty({B}, H.options().dtype(torch::kInt32));
  auto bin = torch::empty({B}, H.options().dtype(torch::kInt32));
  auto lval = torch::empty({B, C}, H.options());
  auto rval = torch::empty({B, C}, H.options());
  if (B > 0)
    hipLaunchKernelGGL(rf_best_split_kernel, dim3((unsigned)B), dim3(256), 0,
                       cur_stream(), H.data_ptr<float>(), F, nb, C,
                       (int)min_leaf, classif ? 1 : 0, gain.data_ptr<float>(),
                       feat.data_ptr<int32_t>(), bin.data_ptr<int32_t>(),
                       lval.data_ptr<float>(), rval.data_ptr<float>());
  return {gain, feat, bin, lval, rval};
}

std::vector<torch::Tensor> kmeans_argmin_kn(torch::Tensor dots, torch::Tensor x_sq,
                                            torch::Tensor c_sq) {
  TORCH_CHECK(dots.is_cuda() && dots.dtype() == torch::kFloat32 && dots.is_contiguous());
  TORCH_CHECK(x_sq.is_contiguous() && c_sq.is_contiguous());
  const int k = (int)dots.size(0);
  const int64_t n = dots.size(1);
  TORCH_CHECK((size_t)k * 4 <= 64 * 1024, "k too large for LDS stage");
  auto labels = torch::empty({n}, dots.options().dtype(torch::kInt32));
  auto min_d = torch::empty({n}, dots.options());
  auto inertia = torch::zeros({1}, dots.options().dtype(torch::kFloat64));
  const unsigned grid = (unsigned)std::min<int64_t>(4096, (n + 255) / 256 + 1);
  hipLaunchKernelGGL(kmeans_argmin_kn_kernel, dim3(grid), dim3(256), (size_t)k * 4,
                     cur_stream(), dots.data_ptr<float>(), x_sq.data_ptr<float>(),
                     c_sq.data_ptr<float>(), n, k, labels.data_ptr<int32_t>(),
                     min_d.data_ptr<float>(), inertia.data_ptr<double>());
  return {labels, min_d, inertia};
}

std::vector<torch::Tensor> kmeans_argmin_nk(torch::Tensor dots, torch::Tensor x_sq,
                                            torch::Tensor c_sq) {
  TORCH_CHECK(dots.is_cuda() && dots.dtype() == torch::kFloat32 && dots.is_contiguous());
  TORCH_CHECK(x_sq.is_contiguous() && c_sq.is_contiguous());
  const int64_t m = dots.size(0);
  const int k = (int)dots.size(1);
  auto labels = torch::empty({m}, dots.options().dtype(torch::kInt32));
  auto min_d = torch::empty({m}, dots.options());
  auto inertia = torch::zeros({1}, dots.options().dtype(torch::kFloat64));
  if (m > 0) {
    const unsigned grid = (unsigned)std::min<int64_t>(16384, (m + 3) / 4);
    hipLaunchKernelGGL(kmeans_argmin_nk_kernel, dim3(grid), dim3(256), 0,
                       cur_stream(), dots.data_ptr<float>(), x_sq.data_ptr<float>(),
                       c_sq.data_ptr<float>(), m, k, labels.data_ptr<int32_t>(),
                       min_d.data_ptr<float>(), inertia.data_ptr<double>());
  }
  return {labels, min_d, inertia};
}

std::vector<torch::Tensor> rf_partition(torch::Tensor node_of_row, torch::Tensor lut,
                                        int64_t B) {
  TORCH_CHECK(node_of_row.is_cuda() && node_of_row.dtype() == torch::kInt64 &&
              node_of_row.is_contiguous());
  TORCH_CHECK(lut.is_cuda() && lut.dtype() == torch::kInt64 && lut.is_contiguous());
  TORCH_CHECK(B >= 1 && B <= 16384, "node batch too large for LDS counts");
  const int64_t n = node_of_row.size(0);
  auto counts = torch::zeros({B}, node_of_row.options().dtype(torch::kInt32));
  const size_t lds1 = (size_t)B * 4;
  const unsigned grid = (unsigned)std::min<int64_t>(1024, (n + 255) / 256 + 1);
  hipLaunchKernelGGL(rf_node_count_kernel, dim3(grid), dim3(256), lds1, cur_stream(),
                     node_of_row.data_ptr<int64_t>(), lut.data_ptr<int64_t>(), n, (int)B,
                     counts.data_ptr<int32_t>());
  auto seg_off = torch::zeros({B + 1}, node_of_row.options().dtype(torch::kInt64));
  seg_off.narrow(0, 1, B).copy_(torch::cumsum(counts, 0));
  auto cursor = seg_off.narrow(0, 0, B).to(torch::kInt32).contiguous();
  auto perm = torch::empty({n}, node_of_row.options().dtype(torch::kInt64));
  const size_t lds2 = (size_t)B * 8;
  hipLaunchKernelGGL(rf_partition_scatter_kernel, dim3(grid), dim3(256), lds2, cur_stream(),
                     node_of_row.data_ptr<int64_t>(), lut.data_ptr<int64_t>(), n, (int)B,
                     cursor.data_ptr<int32_t>(), perm.data_ptr<int64_t>());
  return {perm, seg_off};
}

void rf_reroute(torch::Tensor node_of_row, torch::Tensor lut2, torch::Tensor sfeat,
                torch::Tensor sbin, torch::Tensor lchild, torch::Tensor rchild,
                torch::Tensor Xb, torch::Tensor sample) {
  TORCH_CHECK(node_of_row.is_cuda() && node_of_row.dtype() == torch::kInt64);
  TORCH_CHECK(lut2.dtype() == torch::kInt64 && sfeat.dtype() == torch::kInt32 &&
              sbin.dtype() == torch::kInt32 && lchild.dtype() == torch::kInt64 &&
              rchild.dtype() == torch::kInt64 && Xb.dtype() == torch::kUInt8);
  const int64_t vn = node_of_row.size(0);
  const int64_t n_phys = Xb.size(1);  // [d, n_phys] column-major
  const int64_t d = Xb.size(0);
  const bool has_sample = sample.numel() > 0;
  if (has_sample) TORCH_CHECK(sample.dtype() == torch::kInt32 && sample.is_contiguous());
  const unsigned grid = (unsigned)std::min<int64_t>(2048, (vn + 255) / 256 + 1);
  hipLaunchKernelGGL(rf_reroute_kernel, dim3(grid), dim3(256), 0, cur_stream(),
                     node_of_row.data_ptr<int64_t>(), lut2.data_ptr<int64_t>(),
                     sfeat.data_ptr<int32_t>(), sbin.data_ptr<int32_t>(),
                     lchild.data_ptr<int64_t>(), rchild.data_ptr<int64_t>(),
                     Xb.data_ptr<uint8_t>(),
                     has_sample ? sample.data_ptr<int32_t>() : nullptr, n_phys,
                     vn, d);
}

void knn_merge_topk(torch::Tensor G, torch::Tensor q_sq, torch::Tensor i_sq,
                    int64_t base, torch::Tensor best_d, torch::Tensor best_i) {
  TORCH_CHECK(G.is_cuda() && G.dtype() == torch::kFloat32 && G.is_contiguous());
  TORCH_CHECK(best_d.size(1) == 64 && best_i.size(1) == 64,
              "slot buffers must be [nq, 64]");
  TORCH_CHECK(best_d.is_contiguous() && best_i.is_contiguous());
  TORCH_CHECK(q_sq.is_contiguous() && i_sq.is_contiguous());
  const int64_t nq = G.size(0);
  const int64_t c = G.size(1);
  TORCH_CHECK(i_sq.size(0) == c && q_sq.size(0) == nq && best_d.size(0) == nq);
  hipLaunchKernelGGL(knn_merge_topk_kernel, dim3((unsigned)((nq + 3) / 4)), dim3(256), 0,
                     cur_stream(), G.data_ptr<float>(), q_sq.data_ptr<float>(),
                     i_sq.data_ptr<float>(), nq, c, base, best_d.data_ptr<float>(),
                     best_i.data_ptr<int64_t>());
}

torch::Tensor gather_dists(torch::Tensor A, torch::Tensor B, torch::Tensor cand) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kFloat32 && A.is_contiguous());
  TORCH_CHECK(B.dtype() == torch::kFloat32 && B.is_contiguous());
  TORCH_CHECK(cand.dtype() == torch::kInt64 && cand.is_contiguous());
  const int64_t r = cand.size(0);
  const int64_t m = cand.size(1);
  const int d = (int)A.size(1);
  TORCH_CHECK((int)B.size(1) == d && A.size(0) == r);
  TORCH_CHECK(d <= GD_DMAX, "gather_dists supports d <= 2048");
  auto out = torch::empty({r, m}, A.options());
  if (r > 0 && m > 0)
    hipLaunchKernelGGL(gather_dists_kernel, dim3((unsigned)((r + 3) / 4)), dim3(256), 0,
                       cur_stream(), A.data_ptr<float>(), B.data_ptr<float>(),
                       cand.data_ptr<int64_t>(), r, m, d, out.data_ptr<float>());
  return out;
}

torch::Tensor fil_predict(torch::Tensor X, torch::Tensor feat, torch::Tensor thr,
                          torch::Tensor left, torch::Tensor right, torch::Tensor value,
                          torch::Tensor roots, bool classif) {
  TORCH_CHECK(X.is_cuda() && X.dtype() == torch::kFloat32 && X.is_contiguous());
  TORCH_CHECK(feat.dtype() == torch::kInt32 && left.dtype() == torch::kInt32 &&
              right.dtype() == torch::kInt32 && thr.dtype() == torch::kFloat32 &&
              value.dtype() == torch::kFloat32 && roots.dtype() == torch::kInt64);
  const int64_t n = X.size(0);
  const int d = (int)X.size(1);
  const int T = (int)roots.size(0);
  const int vw = (int)value.size(1);
  TORCH_CHECK(vw <= 16, "fil_predict supports value width <= 16");
  auto out = torch::zeros({n, (int64_t)vw}, X.options());
  if (n > 0 && T > 0) {
    const unsigned grid = (unsigned)std::min<int64_t>(4096, (n + 255) / 256 + 1);
    hipLaunchKernelGGL(fil_predict_kernel, dim3(grid), dim3(256), 0, cur_stream(),
                       X.data_ptr<float>(), feat.data_ptr<int32_t>(),
                       thr.data_ptr<float>(), left.data_ptr<int32_t>(),
                       right.data_ptr<int32_t>(), value.data_ptr<float>(),
                       roots.data_ptr<int64_t>(), n, d, T, vw, classif ? 1 : 0,
                       out.data_ptr<float>());
  }
  return out;
}

torch::Tensor csr_fwd(torch::Tensor indptr, torch::Tensor indices, torch::Tensor vals,
                      torch::Tensor WT) {
  TORCH_CHECK(vals.is_cuda() && vals.dtype() == torch::kFloat32);
  TORCH_CHECK(WT.dtype() == torch::kFloat32 && WT.is_contiguous());
  const int64_t n = indptr.size(0) - 1;
  const int C = (int)WT.size(1);
  TORCH_CHECK(C <= 32, "csr_fwd supports C <= 32");
  auto scores = torch::empty({n, (int64_t)C}, vals.options());
  const unsigned grid = (unsigned)std::min<int64_t>(4096, (n + 255) / 256 + 1);
  if (indptr.dtype() == torch::kInt32) {
    hipLaunchKernelGGL(csr_fwd_kernel<int32_t>, dim3(grid), dim3(256), 0, cur_stream(),
                       indptr.data_ptr<int32_t>(), indices.data_ptr<int32_t>(),
                       vals.data_ptr<float>(), WT.data_ptr<float>(), n, C,
                       scores.data_ptr<float>());
  } else {
    hipLaunchKernelGGL(csr_fwd_kernel<int64_t>, dim3(grid), dim3(256), 0, cur_stream(),
                       indptr.data_ptr<int64_t>(), indices.data_ptr<int64_t>(),
                       vals.data_ptr<float>(), WT.data_ptr<float>(), n, C,
                       scores.data_ptr<float>());
  }
  return scores;
}

torch::Tensor csr_grad(torch::Tensor indptr, torch::Tensor indices, torch::Tensor vals,
                       torch::Tensor resid, int64_t d) {
  TORCH_CHECK(vals.is_cuda() && vals.dtype() == torch::kFloat32);
  TORCH_CHECK(resid.dtype() == torch::kFloat32 && resid.is_contiguous());
  const int64_t n = indptr.size(0) - 1;
  const int C = (int)resid.size(1);
  TORCH_CHECK(C <= 32, "csr_grad supports C <= 32");
  auto grad = torch::zeros({(int64_t)C, d}, vals.options());
  const unsigned grid = (unsigned)std::min<int64_t>(4096, (n + 255) / 256 + 1);
  const int64_t cd = (int64_t)C * d;
  const int use_lds = (cd * 4 <= 120 * 1024) ? 1 : 0;
  const size_t lds = use_lds ? (size_t)cd * 4 : 0;
  if (indptr.dtype() == torch::kInt32) {
    hipLaunchKernelGGL(csr_grad_kernel<int32_t>, dim3(grid), dim3(256), lds, cur_stream(),
                       indptr.data_ptr<int32_t>(), indices.data_ptr<int32_t>(),
                       vals.data_ptr<float>(), resid.data_ptr<float>(), n, d, C, use_lds,
                       grad.data_ptr<float>());
  } else {
    hipLaunchKernelGGL(csr_grad_kernel<int64_t>, dim3(grid), dim3(256), lds, cur_stream(),
                       indptr.data_ptr<int64_t>(), indices.data_ptr<int64_t>(),
                       vals.data_ptr<float>(), resid.data_ptr<float>(), n, d, C, use_lds,
                       grad.data_ptr<float>());
  }
  return grad;
}

torch::Tensor csr_col_moments(torch::Tensor indices, torch::Tensor vals, int64_t d) {
  TORCH_CHECK(vals.is_cuda() && vals.dtype() == torch::kFloat32);
  TORCH_CHECK(d * 16 <= 160 * 1024, "csr_col_moments: d too large for LDS");
  const int64_t nnz = vals.size(0);
  auto out = torch::zeros({2, d}, vals.options().dtype(torch::kFloat64));
  const unsigned grid = (unsigned)std::min<int64_t>(2048, (nnz + 255) / 256 + 1);
  const size_t lds = (size_t)d * 16;
  if (indices.dtype() == torch::kInt32) {
    hipLaunchKernelGGL(csr_col_moments_kernel<int32_t>, dim3(grid), dim3(256), lds,
                       cur_stream(), indices.data_ptr<int32_t>(),
                       vals.data_ptr<float>(), nnz, d, out.data_ptr<double>());
  } else {
    hipLaunchKernelGGL(csr_col_moments_kernel<int64_t>, dim3(grid), dim3(256), lds,
                       cur_stream(), indices.data_ptr<int64_t>(),
                       vals.data_ptr<float>(), nnz, d, out.data_ptr<double>());
  }
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fil_predict", &fil_predict, "forest inference (FIL-style traversal)");
  m.def("csr_fwd", &csr_fwd, "CSR scores = A @ WT (thread per row)");
  m.def("csr_grad", &csr_grad, "CSR grad = (A^T R)^T by column scatter");
  m.def("csr_col_moments", &csr_col_moments, "per-column sum/sumsq of CSR values");
  m.def("kmeans_assign", &kmeans_assign, "fused MFMA distance + argmin");
  m.def("kmeans_argmin_kn", &kmeans_argmin_kn, "argmin epilogue over a [k,n] GEMM dot block");
  m.def("kmeans_argmin_nk", &kmeans_argmin_nk, "argmin epilogue over a row-major [m,k] GEMM dot block");
  m.def("label_accumulate", &label_accumulate, "per-center sum/count scatter");
  m.def("gram_f32", &gram_f32, "A^T A via MFMA f32");
  m.def("softmax_residual_loss", &softmax_residual_loss, "fused softmax residual + loss");
  m.def("knn_select", &knn_select, "fused MFMA distance + in-LDS top-k");
  m.def("rf_histogram", &rf_histogram, "LDS-privatized RF split histograms");
  m.def("rf_histogram_fw", &rf_histogram_fw, "feature-wide RF histograms (lanes over sorted features)");
  m.def("rf_best_split", &rf_best_split, "fused RF gain scan + block-best reduce");
  m.def("dbscan_sweep", &dbscan_sweep, "fused eps-neighborhood count / min-core-label pass");
  m.def("umap_sgd", &umap_sgd, "edge-sampled UMAP SGD (all epochs, Hogwild)");
  m.def("rf_partition", &rf_partition, "counting-sort rows by node -> (perm, seg_off)");
  m.def("rf_reroute", &rf_reroute, "one-pass row-to-child reassignment");
  m.def("knn_merge_topk", &knn_merge_topk, "running top-k merge over a GEMM dot block");
  m.def("gather_dists", &gather_dists, "gathered row-vs-candidates squared distances");
  m.attr("_is_hip") = true;
}
