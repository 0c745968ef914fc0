// ALS kernels for MI355X (gfx950, CDNA4).
//
// Replaces the reference's Spark-MLlib ALS normal-equation construction +
// per-row Cholesky solve (reference call sites: examples/scala-parallel-
// recommendation/.../ALSAlgorithm.scala:75-86 `ALS.train`; similarproduct
// `ALSAlgorithm.scala:130-136` trainImplicit).
//
// Design (MI355X-first, not a port):
//  - One 256-thread workgroup (4 wave64) per output row (user or item).
//  - The row's rated factor vectors are staged through LDS in chunks;
//    each thread register-tiles a TMxTM block of the FxF Gramian
//    (fp32 accumulate — CDNA4 has no fp32 MFMA; the Gramian pass is
//    VALU fp32 with ds_read_b128 vector LDS reads).
//  - Cholesky factorization + triangular solves run in LDS on the same
//    workgroup — the Gramian never touches HBM (saves 16 KB/row of
//    HBM traffic at f=64).
//  - Implicit mode (Hu-Koren): A_u = YtY + sum_i alpha*r * y_i y_i^T + l*I,
//    b_u = sum_i (1 + alpha*r) y_i, with YtY precomputed once per
//    half-iteration (a plain library GEMM).
//  - Explicit mode (ALS-WR like MLlib): A_u = sum y y^T + l*nnz_u*I,
//    b_u = sum r*y.
//
// F (rank) is a template parameter in {16, 32, 64, 128}; callers pad.

#include <hip/hip_runtime.h>

#define CHUNK 16  // rated rows staged per LDS pass

__device__ __forceinline__ void wave_sync() {
  __builtin_amdgcn_wave_barrier();
}

typedef __attribute__((ext_vector_type(4))) float f32x4_t;


template <int F>
__global__ __launch_bounds__(256) void als_solve_kernel(
    const long long* __restrict__ indptr,   // n_rows+1
    const int* __restrict__ indices,        // nnz (column ids into Y)
    const float* __restrict__ values,       // nnz
    const float* __restrict__ Y,            // n_cols x F (fixed side)
    const float* __restrict__ YtY,          // F x F or nullptr
    float* __restrict__ X,                  // n_rows x F (output)
    int n_rows,
    float lambda,
    float alpha,
    int implicit_mode,                      // 1 = Hu-Koren implicit
    int wr_scale,                           // 1 = scale lambda by nnz (ALS-WR)
    int skip_below)                         // skip rows with nnz <= this
{
  constexpr int TM = 4;                       // thread tile edge
  constexpr int TILES = (F / TM) * (F / TM);  // tiles covering FxF
  constexpr int TPT = (TILES + 255) / 256;    // tiles per thread
  constexpr int NB = 16;                      // Cholesky panel width
  static_assert(F % TM == 0, "F must be divisible by 4");
  static_assert(F % NB == 0, "F must be divisible by the panel width");

  // +1 padding on the staging buffer breaks the F-stride bank conflict
  // (MI355X LDS = 32 banks x 4 B; F=64 floats stride = same-bank).
  __shared__ float ys[CHUNK][F + 1];
  __shared__ float ws_a[CHUNK];  // Gramian weight per staged row
  __shared__ float ws_b[CHUNK];  // b-vector weight per staged row
  // As rows padded to F+4: row starts stay 16-byte aligned, so the
  // rank-NB trailing update can read L panels as float4 quads (the same
  // ds_read_b128 lesson as the wave kernel's pivot columns).
  __shared__ float As[F][F + 4];
  __shared__ float bs[F];
  __shared__ float dinv_all[F];  // 1/L[k][k], filled during factorization

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  for (long long row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const long long start = indptr[row];
    const long long end = indptr[row + 1];
    const int nnz = (int)(end - start);
    if (nnz <= skip_below) continue;  // Woodbury kernel owns these rows

    // register accumulators: TPT tiles of TMxTM
    float acc[TPT][TM][TM];
#pragma unroll
    for (int t = 0; t < TPT; ++t)
#pragma unroll
      for (int i = 0; i < TM; ++i)
#pragma unroll
        for (int j = 0; j < TM; ++j) acc[t][i][j] = 0.f;

    // zero b
    for (int i = tid; i < F; i += 256) bs[i] = 0.f;

    for (int base = 0; base < nnz; base += CHUNK) {
      const int cn = min(CHUNK, nnz - base);
      __syncthreads();
      // stage cn factor rows: thread t loads element (t % F) of row (t / F)
      for (int e = tid; e < cn * F; e += 256) {
        const int c = e / F;
        const int k = e % F;
        const int col = indices[start + base + c];
        ys[c][k] = Y[(long long)col * F + k];
      }
      if (tid < cn) {
        const float r = values[start + base + tid];
        if (implicit_mode) {
          ws_a[tid] = alpha * r;
          ws_b[tid] = 1.f + alpha * r;
        } else {
          ws_a[tid] = 1.f;
          ws_b[tid] = r;
        }
      }
      __syncthreads();

      // b accumulation (threads 0..F-1)
      if (tid < F) {
        float bacc = 0.f;
        for (int c = 0; c < cn; ++c) bacc += ws_b[c] * ys[c][tid];
        bs[tid] += bacc;
      }

      // Gramian accumulation: each thread's TMxTM tiles
#pragma unroll
      for (int t = 0; t < TPT; ++t) {
        const int tile = tid + t * 256;
        if (TILES >= 256 || tile < TILES) {
          const int ti = (tile / (F / TM)) * TM;
          const int tj = (tile % (F / TM)) * TM;
          for (int c = 0; c < cn; ++c) {
            const float w = ws_a[c];
            float ya[TM], yb[TM];
#pragma unroll
            for (int i = 0; i < TM; ++i) ya[i] = ys[c][ti + i];
#pragma unroll
            for (int j = 0; j < TM; ++j) yb[j] = ys[c][tj + j];
#pragma unroll
            for (int i = 0; i < TM; ++i)
#pragma unroll
              for (int j = 0; j < TM; ++j)
                acc[t][i][j] = fmaf(w * ya[i], yb[j], acc[t][i][j]);
          }
        }
      }
    }
    __syncthreads();

    // write accumulators to LDS A (+ YtY base for implicit)
#pragma unroll
    for (int t = 0; t < TPT; ++t) {
      const int tile = tid + t * 256;
      if (TILES >= 256 || tile < TILES) {
        const int ti = (tile / (F / TM)) * TM;
        const int tj = (tile % (F / TM)) * TM;
#pragma unroll
        for (int i = 0; i < TM; ++i)
#pragma unroll
          for (int j = 0; j < TM; ++j) {
            float v = acc[t][i][j];
            if (YtY != nullptr) v += YtY[(ti + i) * F + (tj + j)];
            As[ti + i][tj + j] = v;
          }
      }
    }
    __syncthreads();
    // regularization diagonal
    if (tid < F) {
      const float reg = wr_scale ? lambda * (float)nnz : lambda;
      As[tid][tid] += reg;
    }
    __syncthreads();

    // ---- in-LDS blocked Cholesky (NB-wide panels, lower triangle) ----
    // The unblocked right-looking loop ran 3 __syncthreads per pivot
    // (3F = 384 s_barriers at F=128) and measured ~450 us/row of nearly
    // pure barrier/serialization cost (microbench: time flat in nnz).
    // Blocked form: per panel, (1) wave 0 factors the NBxNB diagonal
    // block alone — intra-wave lockstep + wave_sync() instead of
    // s_barrier, the same free-running idiom as the wave kernel above;
    // (2) the panel below it is a triangular solve with INDEPENDENT
    // rows — one row per thread, no barriers; (3) a rank-NB trailing
    // update amortizes one barrier over NB pivots. 3 s_barriers per
    // panel = 24 at F=128 instead of 384.
    for (int k0 = 0; k0 < F; k0 += NB) {
      if (wave == 0) {
        // (1) diagonal block, lanes 0..NB-1 of wave 0 (lane i = row i)
        for (int kk = 0; kk < NB; ++kk) {
          const int kq = k0 + kk;
          wave_sync();
          float d = As[kq][kq];
          d = d > 0.f ? sqrtf(d) : 1e-20f;
          const float dinv = 1.f / d;     // redundant on all lanes
          if (lane == kk) {
            As[kq][kq] = d;
            dinv_all[kq] = dinv;
          }
          if (lane > kk && lane < NB) As[k0 + lane][kq] *= dinv;
          wave_sync();
          if (lane > kk && lane < NB) {
            const float lik = As[k0 + lane][kq];
            for (int j = kk + 1; j <= lane; ++j)
              As[k0 + lane][k0 + j] =
                  fmaf(-lik, As[k0 + j][kq], As[k0 + lane][k0 + j]);
          }
        }
      }
      __syncthreads();
      if (k0 + NB >= F) break;  // no panel below the last block
      // (2) panel solve: row i of A[k0+NB.., k0..k0+NB) <- row * L_bb^-T.
      // Rows are independent: one thread per row, registers w[NB]
      // literal-indexed via full unroll (runtime indices would spill).
      for (int i = k0 + NB + tid; i < F; i += 256) {
        float w[NB];
#pragma unroll
        for (int m = 0; m < NB; ++m) w[m] = As[i][k0 + m];
#pragma unroll
        for (int j = 0; j < NB; ++j) {
          float s = w[j];
#pragma unroll
          for (int m = 0; m < NB; ++m)
            if (m < j) s = fmaf(-w[m], As[k0 + j][k0 + m], s);
          w[j] = s * dinv_all[k0 + j];
        }
#pragma unroll
        for (int m = 0; m < NB; ++m) As[i][k0 + m] = w[m];
      }
      __syncthreads();
      // (3) rank-NB trailing update: A[i][j] -= sum_m L[i][m] L[j][m].
      // 2x2 tiles, panel rows read as float4 quads (rows are 16B-aligned
      // and k0 is a multiple of NB=16): 4 ds_read_b128 per 4 m-terms of
      // 4 elements instead of 32 scalar ds_read_b32.
      const int rem = F - k0 - NB;
      const int remt = rem >> 1;
      for (int e = tid; e < remt * remt; e += 256) {
        const int i = k0 + NB + 2 * (e / remt);
        const int j = k0 + NB + 2 * (e % remt);
        if (j <= i) {
          float s00 = As[i][j];
          float s01 = As[i][j + 1];
          float s10 = As[i + 1][j];
          float s11 = As[i + 1][j + 1];
#pragma unroll
          for (int m = 0; m < NB; m += 4) {
            const f32x4_t a0 =
                *reinterpret_cast<const f32x4_t*>(&As[i][k0 + m]);
            const f32x4_t a1 =
                *reinterpret_cast<const f32x4_t*>(&As[i + 1][k0 + m]);
            const f32x4_t b0 =
                *reinterpret_cast<const f32x4_t*>(&As[j][k0 + m]);
            const f32x4_t b1 =
                *reinterpret_cast<const f32x4_t*>(&As[j + 1][k0 + m]);
#pragma unroll
            for (int q = 0; q < 4; ++q) {
              s00 = fmaf(-a0[q], b0[q], s00);
              s01 = fmaf(-a0[q], b1[q], s01);
              s10 = fmaf(-a1[q], b0[q], s10);
              s11 = fmaf(-a1[q], b1[q], s11);
            }
          }
          As[i][j] = s00;
          if (j + 1 <= i) As[i][j + 1] = s01;
          As[i + 1][j] = s10;
          As[i + 1][j + 1] = s11;
        }
      }
      __syncthreads();
    }

    // ---- forward solve L z = b, blocked the same way ----
    for (int k0 = 0; k0 < F; k0 += NB) {
      if (wave == 0) {
        for (int kk = 0; kk < NB; ++kk) {
          wave_sync();
          const float zk = bs[k0 + kk] * dinv_all[k0 + kk];
          if (lane == kk) bs[k0 + kk] = zk;
          if (lane > kk && lane < NB)
            bs[k0 + lane] = fmaf(-As[k0 + lane][k0 + kk], zk,
                                 bs[k0 + lane]);
        }
      }
      __syncthreads();
      for (int j = k0 + NB + tid; j < F; j += 256) {
        float s = bs[j];
#pragma unroll
        for (int m = 0; m < NB; ++m)
          s = fmaf(-As[j][k0 + m], bs[k0 + m], s);
        bs[j] = s;
      }
      __syncthreads();
    }
    // ---- back solve L^T x = z ----
    for (int k0 = F - NB; k0 >= 0; k0 -= NB) {
      if (wave == 0) {
        for (int kk = NB - 1; kk >= 0; --kk) {
          wave_sync();
          const float xk = bs[k0 + kk] * dinv_all[k0 + kk];
          if (lane == kk) bs[k0 + kk] = xk;
          if (lane < kk)
            bs[k0 + lane] = fmaf(-As[k0 + kk][k0 + lane], xk,
                                 bs[k0 + lane]);
        }
      }
      __syncthreads();
      for (int j = tid; j < k0; j += 256) {
        float s = bs[j];
#pragma unroll
        for (int m = 0; m < NB; ++m)
          s = fmaf(-As[k0 + m][j], bs[k0 + m], s);
        bs[j] = s;
      }
      __syncthreads();
    }

    if (tid < F) X[row * (long long)F + tid] = bs[tid];
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// Wave-per-row solver (F <= 64) — the fast path for mid-size rows.
//
// The workgroup kernel above spends most of its time in __syncthreads-heavy
// in-LDS Cholesky (measured: iteration time nearly flat in nnz — fixed
// per-row solve cost dominates). This version assigns one 64-lane wave per
// output row (2 waves per workgroup), keeps the Gramian in REGISTERS during
// accumulation, and factorizes with the computed L columns persisted in LDS:
//   - lane j owns row j of the FxF Gramian: acc[F] VGPRs, literal-indexed
//     via fully unrolled loops (runtime-indexed register arrays spill).
//   - Cholesky: per step k the scaled pivot column is written to a per-wave
//     LDS matrix Lc[k][*] (broadcast reads are conflict-free); no
//     __syncthreads — waves are free-running, intra-wave LDS ordering is
//     enforced by lockstep execution + wave_barrier. Persisting EVERY column
//     (rather than capturing the transpose into upper-triangle registers)
//     keeps register live ranges short: a first cut that captured L^T into
//     dead acc[] registers ballooned to 256 VGPRs + 3 KB scratch spill and
//     ran 2x slower than the workgroup kernel (rocprof: private_segment
//     3020 B).
//   - back solve reads L^T straight from the persisted LDS columns.
//   - Gramian: per rated item, each lane loads its element of y (coalesced
//     256 B line), stages it through LDS, and rank-1-updates its register
//     row from broadcast float4 reads; the next item's global load is
//     issued before the current item's math (double-buffered LDS).
// ---------------------------------------------------------------------------

// BF16G: the Gramian runs on the MATRIX CORES — the north-star's
// "ALS factor-update Gramian ... on MFMA". Items are staged TRANSPOSED
// as bf16 (ysT[feature][item], rows scaled by sqrt(w_a) so the product
// is a plain S^T S), and because S^T S is symmetric the A- and
// B-operand fragments are the SAME four b128 loads per 32-item stripe:
// 4 ds_read_b128 feed 16 v_mfma_f32_16x16x32_bf16 accumulating the
// full 64x64 normal matrix in fp32. One LDS pass through Lc
// redistributes D's fragment layout (col = lane&15) into the solver's
// lane-as-column acc[F]. Opt-in via PIO_ALS_STAGE_BF16 (same umbrella
// as the Woodbury bf16 staging; b and the Cholesky stay fp32).
template <int F, bool BF16G = false>
__global__ __launch_bounds__(128) void als_solve_wave_kernel(
    const long long* __restrict__ indptr,
    const int* __restrict__ indices,
    const float* __restrict__ values,
    const float* __restrict__ Y,
    const float* __restrict__ YtY,
    float* __restrict__ X,
    int n_rows,
    float lambda,
    float alpha,
    int implicit_mode,
    int wr_scale,
    int skip_below,
    int skip_above)   // rows with nnz > this go to the workgroup kernel
{
  static_assert(F <= 64, "wave kernel supports rank <= 64");
  static_assert(!BF16G || F == 64, "MFMA Gramian variant needs F=64");
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  constexpr int PD = 4;  // global-load pipeline depth (latency hiding).
  // PD=8 and PD=12 MEASURED SLOWER (round 2: nnz=200 23.2->23.9/26.8 ms
  // per 0.2M rows) despite LDS capping occupancy either way — the extra
  // in-flight registers hurt scheduling more than the added latency
  // coverage helps; the per-item LDS broadcast + two wave_syncs, not
  // HBM latency, bound this loop
  constexpr int LP = F + 4;  // Lc row stride: 16B-aligned for b128 reads
  __shared__ float ys[2][PD][F];       // staged y slots per wave
  // BF16G: transposed bf16 stripe, rows padded to 40 (80 B, 16B-mult);
  // granule-XOR (row&3) spreads the strided column writes
  constexpr int TP = 40;
  __shared__ unsigned short ysT[2][BF16G ? F : 1][BF16G ? TP : 1];
  __shared__ float Lc[2][F][LP];       // persisted L columns: Lc[w][k][j] = L[j][k]

  for (long long row = (long long)blockIdx.x * 2 + wave; row < n_rows;
       row += (long long)gridDim.x * 2) {
    const long long start = indptr[row];
    const int nnz = (int)(indptr[row + 1] - start);
    if (nnz <= skip_below || nnz > skip_above)
      continue;  // Woodbury / workgroup kernels own these rows

    float acc[F];
#pragma unroll
    for (int m = 0; m < F; ++m) acc[m] = 0.f;
    float b_reg = 0.f;

    if constexpr (BF16G) {
      // ---- MFMA Gramian (see the kernel comment) ----
      typedef __attribute__((ext_vector_type(8))) unsigned short bf16x8g;
      typedef __attribute__((ext_vector_type(4))) float f32x4g;
      const int lg = lane >> 4;   // fragment k-granule
      const int lq = lane & 15;   // fragment row-in-block
      f32x4g accm[4][4];
#pragma unroll
      for (int ti = 0; ti < 4; ++ti)
#pragma unroll
        for (int tj = 0; tj < 4; ++tj)
          accm[ti][tj] = f32x4g{0.f, 0.f, 0.f, 0.f};
      // 4-deep pipelined item loads (per item: one coalesced f32 row)
      float yn2[PD], wa2[PD], wb2[PD];
#pragma unroll
      for (int p = 0; p < PD; ++p) {
        if (p < nnz) {
          const int col = indices[start + p];
          yn2[p] = lane < F ? Y[(long long)col * F + lane] : 0.f;
          const float v = values[start + p];
          wa2[p] = implicit_mode ? alpha * v : 1.f;
          wb2[p] = implicit_mode ? 1.f + alpha * v : v;
        }
      }
      for (int c0 = 0; c0 < nnz; c0 += 32) {
        for (int cm = 0; cm < 32; ++cm) {
          const int c = c0 + cm;
          const int slot = c & (PD - 1);
          float sy = 0.f;
          if (c < nnz) {
            const float wa = wa2[slot];
            const float sc = sqrtf(wa > 1e-12f ? wa : 1e-12f);
            const float ycur = yn2[slot];
            sy = sc * ycur;
            b_reg = fmaf(wb2[slot], ycur, b_reg);   // b stays fp32-exact
            if (c + PD < nnz) {
              const int ncol = indices[start + c + PD];
              yn2[slot] = lane < F ? Y[(long long)ncol * F + lane] : 0.f;
              const float nv = values[start + c + PD];
              wa2[slot] = implicit_mode ? alpha * nv : 1.f;
              wb2[slot] = implicit_mode ? 1.f + alpha * nv : nv;
            }
          }
          // transposed bf16 store: ysT[feature=lane][item cm], granule
          // (cm>>3) XOR'd with (row&3)
          const unsigned short hb =
              (unsigned short)(__float_as_uint(sy) >> 16);
          const int gcol = (((cm >> 3) ^ (lane & 3)) << 3) | (cm & 7);
          ysT[wave][lane][gcol] = hb;
        }
        wave_sync();
        bf16x8g frag[4];
#pragma unroll
        for (int t = 0; t < 4; ++t) {
          const int row = t * 16 + lq;
          const int g = lg ^ (row & 3);
          frag[t] = *reinterpret_cast<const bf16x8g*>(
              &ysT[wave][row][g << 3]);
        }
#pragma unroll
        for (int ti = 0; ti < 4; ++ti)
#pragma unroll
          for (int tj = 0; tj < 4; ++tj)
            accm[ti][tj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                frag[ti], frag[tj], accm[ti][tj], 0, 0, 0);
        wave_sync();
      }
      // redistribute D (col = lane&15, row = (lane>>4)*4 + r) into the
      // solver's lane-as-column acc[F] through Lc (free until Cholesky)
#pragma unroll
      for (int ti = 0; ti < 4; ++ti)
#pragma unroll
        for (int tj = 0; tj < 4; ++tj)
#pragma unroll
          for (int r = 0; r < 4; ++r)
            Lc[wave][ti * 16 + lg * 4 + r][tj * 16 + lq] =
                accm[ti][tj][r];
      wave_sync();
#pragma unroll
      for (int m = 0; m < F; ++m) acc[m] = Lc[wave][m][lane];
      wave_sync();
    } else {
    // ---- Gramian accumulation (register row, LDS broadcast) ----
    // PD-deep load pipeline: a first cut double-buffered one item at a
    // time and was latency-bound (~940 cycles/item measured at nnz=40);
    // keeping PD row loads in flight divides the exposed HBM latency.
    float yn[PD], vn[PD];
#pragma unroll
    for (int p = 0; p < PD; ++p) {
      if (p < nnz) {
        const int col = indices[start + p];
        yn[p] = lane < F ? Y[(long long)col * F + lane] : 0.f;
        vn[p] = values[start + p];
      }
    }
    for (int c0 = 0; c0 < nnz; c0 += PD) {
#pragma unroll
      for (int p = 0; p < PD; ++p) {
        const int c = c0 + p;
        if (c >= nnz) break;
        const float ycur = yn[p];
        const float vcur = vn[p];
        if (lane < F) ys[wave][p][lane] = ycur;
        wave_sync();
        if (c + PD < nnz) {  // refill this slot PD items ahead
          const int ncol = indices[start + c + PD];
          if (lane < F) yn[p] = Y[(long long)ncol * F + lane];
          vn[p] = values[start + c + PD];
        }
        float w_a, w_b;
        if (implicit_mode) {
          w_a = alpha * vcur;
          w_b = 1.f + alpha * vcur;
        } else {
          w_a = 1.f;
          w_b = vcur;
        }
        b_reg = fmaf(w_b, ycur, b_reg);
        const float wyj = w_a * ycur;
#pragma unroll
        for (int q = 0; q < F / 4; ++q) {
          const float4 y4 =
              *reinterpret_cast<const float4*>(&ys[wave][p][4 * q]);
          acc[4 * q + 0] = fmaf(wyj, y4.x, acc[4 * q + 0]);
          acc[4 * q + 1] = fmaf(wyj, y4.y, acc[4 * q + 1]);
          acc[4 * q + 2] = fmaf(wyj, y4.z, acc[4 * q + 2]);
          acc[4 * q + 3] = fmaf(wyj, y4.w, acc[4 * q + 3]);
        }
        wave_sync();
      }
    }

    }

    // ---- YtY base (implicit) + regularization ----
    if (YtY != nullptr && lane < F) {
#pragma unroll
      for (int q = 0; q < F / 4; ++q) {
        const float4 t4 = *reinterpret_cast<const float4*>(
            &YtY[(long long)lane * F + 4 * q]);
        acc[4 * q + 0] += t4.x;
        acc[4 * q + 1] += t4.y;
        acc[4 * q + 2] += t4.z;
        acc[4 * q + 3] += t4.w;
      }
    }
    {
      const float reg = wr_scale ? lambda * (float)nnz : lambda;
#pragma unroll
      for (int m = 0; m < F; ++m)
        if (m == lane) acc[m] += reg;
    }

    // ---- Cholesky (k-loop unrolled so acc[] stays in registers; each
    //      finished column goes to LDS and acc[k] dies at step k) ----
#pragma unroll
    for (int k = 0; k < F; ++k) {
      float lkk = __shfl(acc[k], k);
      // clamp so a singular (all-zero) row divides by a tiny nonzero pivot
      // instead of 0 (empty rows must yield x = 0, not NaN)
      lkk = lkk > 1e-30f ? lkk : 1e-30f;
      const float dinv = rsqrtf(lkk);
      const float ljk = lane > k ? acc[k] * dinv
                                 : (lane == k ? lkk * dinv : 0.f);
      // persist column k: Lc[k][j] = L[j][k]. Guard: for F < 64 lanes
      // >= F would write past the row into the next column. acc[k] is
      // dead after this step — keep L[lane][k] in it so the forward
      // solve runs from registers (round 2: the scalar Lc reads in the
      // substitutions were the flagged ds_read_b32 round trips).
      if (lane < F) Lc[wave][k][lane] = ljk;
      acc[k] = ljk;
      wave_sync();
      // trailing update reading the pivot column in b128 QUADS: the
      // scalar version compiled to ds_read_b32 + s_waitcnt lgkmcnt(0)
      // per element — a full LDS round trip exposed F-k times per step
      // (seen in the ISA; the fixed per-row solve cost dominated the
      // kernel). One quad read amortizes the wait over 4 fmas.
#pragma unroll
      for (int m4 = (k + 1) & ~3; m4 < F; m4 += 4) {
        const float4 lq =
            *reinterpret_cast<const float4*>(&Lc[wave][k][m4]);
        if (lane > k) {
          if (m4 + 0 > k) acc[m4 + 0] = fmaf(-ljk, lq.x, acc[m4 + 0]);
          if (m4 + 1 > k) acc[m4 + 1] = fmaf(-ljk, lq.y, acc[m4 + 1]);
          if (m4 + 2 > k) acc[m4 + 2] = fmaf(-ljk, lq.z, acc[m4 + 2]);
          if (m4 + 3 > k) acc[m4 + 3] = fmaf(-ljk, lq.w, acc[m4 + 3]);
        }
      }
      wave_sync();
    }

    // ---- forward solve L z = b — pure register/shfl chain: lane's row
    //      L[lane][k] lives in acc[k], the pivot L[k][k] is lane k's
    //      acc[k] (no LDS round trips on the serial chain) ----
#pragma unroll
    for (int k = 0; k < F; ++k) {
      const float zk = __shfl(b_reg, k) / __shfl(acc[k], k);
      if (lane == k) b_reg = zk;
      else if (lane > k) b_reg = fmaf(-acc[k], zk, b_reg);
    }
    // ---- back solve L^T x = z: lane j needs L[k][j] = Lc[j][k], which
    //      is CONSECUTIVE over k at fixed row j — read b128 quads so one
    //      LDS round trip covers 4 serial steps ----
    {
      float4 lq = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int k = F - 1; k >= 0; --k) {
        if ((k & 3) == 3 && lane < F)
          lq = *reinterpret_cast<const float4*>(&Lc[wave][lane][k & ~3]);
        const float xk = __shfl(b_reg, k) / __shfl(acc[k], k);
        if (lane == k) b_reg = xk;
        else if (lane < k) {
          const float lv = (k & 3) == 0 ? lq.x
                           : (k & 3) == 1 ? lq.y
                           : (k & 3) == 2 ? lq.z : lq.w;
          b_reg = fmaf(-lv, xk, b_reg);
        }
      }
    }

    if (lane < F) X[row * (long long)F + lane] = b_reg;
  }
}

// ---------------------------------------------------------------------------
// Woodbury solver — rows with nnz <= WOODBURY_MAX_NNZ (the common case for
// recommendation data: the bench config has nnz/row = 20, rank = 64).
//
// Instead of building + factorizing the FxF normal matrix per row, exploit
// that only nnz rank-1 terms differ from a FIXED base B:
//   implicit (Hu-Koren):  A_u = B + U C U^T,  B = Y^T Y + lambda I = L L^T,
//     U = [y_i] (F x n), C = diag(alpha r_i), b_u = U c2, c2_i = 1+alpha r_i.
//     Push-through identity:  x_u = B^-1 U (I + C G)^-1 c2,
//     with G = U^T B^-1 U.  Using the WHITENED factors V = Y L^-T
//     (precomputed once per half-iteration — one host-side trsm):
//       G = V_u . V_u^T   (v_i = L^-1 y_i, so v_i.v_j = y_i^T B^-1 y_j)
//     Symmetrized with D = C^(1/2): solve (I + D G D) t = c2 / d, s = d t,
//     and emit z_u = sum_i s_i v_i; the host maps X = Z L^-1 for ALL rows
//     in one triangular-solve GEMM (x = L^-T z).  Only ONE staged factor
//     matrix per row — half the HBM traffic and LDS of the W = B^-1 Y
//     formulation, doubling resident waves.
//   explicit (ALS-WR):  A_u = U U^T + reg I, b_u = U r:
//     x_u = U (G + reg I)^-1 r with G = Y_u . Y_u^T — same kernel body
//     with Y staged instead of V and X emitted directly (no host solve).
//
// Cost per row: n^2/2 F MACs for G + n^3/3 solve — ~5x fewer FLOPs than
// the FxF Cholesky at n=20, F=64, with NO long serial dependency chains.
// One wave per row, 2 waves per workgroup; rows with nnz > WOODBURY_MAX_NNZ
// are skipped here and handled by the wave/workgroup kernels.
// LDS rows padded to F+4 floats so G-dot reads from row i (stride 68) land
// in different bank groups per lane.
// ---------------------------------------------------------------------------

#define WOODBURY_MAX_NNZ 32


// Two NW instantiations: rows with nnz <= 24 run the NW=24 variant whose
// smaller LDS footprint (9.7 vs 13.3 KB/wave) fits 16 waves/CU instead of
// 12 and whose unrolled M-solve is 25% shorter — the phase probe showed
// the kernel issue-bound at its occupancy cap with the M-solve at 66%.
// BF16S: stage the factor rows (V in implicit mode) as bf16 — halves the
// occupancy-capping Yl LDS buffer AND the stage HBM bytes. The round-2
// TCC probe showed this kernel latency-bound (SQ_WAIT:BUSY 19:1) at only
// ~310 GB/s effective, so resident waves are the lever; the bf16
// numerics study (profiles/bf16_numerics_study.txt) bounds the staged-
// factor rounding at ~1e-5 relative ALS objective. Opt-in via
// PIO_ALS_STAGE_BF16=1 (the V pointer then carries bf16 data).
template <int F, int NW, int NLO, bool BF16S = false>
__global__ __launch_bounds__(128, 6) void als_woodbury_kernel(
    const long long* __restrict__ indptr,
    const int* __restrict__ indices,
    const float* __restrict__ values,
    const float* __restrict__ Y,     // item factors (explicit mode)
    const float* __restrict__ V,     // whitened factors Y L^-T (implicit)
    float* __restrict__ X,           // X (explicit) / Z (implicit)
    int n_rows,
    float lambda,
    float alpha,
    int implicit_mode,
    int wr_scale,
    unsigned long long* prof)        // optional [5]: phase wall-clock sums
                                     // (stage, G, solve, emit, samples) —
                                     // ATT is unavailable on this image,
                                     // so the kernel self-times sampled
                                     // rows with wall_clock64()
{
  constexpr int FP = F + 4;    // fp32 row stride (floats)
  constexpr int FPH = F + 8;   // bf16 row stride (ushorts; 16B multiple)
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  __shared__ float Yl[2][NW][BF16S ? (FPH / 2) : FP];  // staged rows
  constexpr int MP = NW + 4;  // M row stride: 16B-aligned for b128 reads
  __shared__ float M[2][NW][MP];       // I+DGD (implicit) / G+regI (explicit)
  __shared__ float tv[2][NW];          // col ids, then rhs, then solution
  __shared__ float dv[2][NW];          // D diagonal (implicit)

  float* yl = &Yl[wave][0][0];
  unsigned short* ylh = reinterpret_cast<unsigned short*>(yl);
  const float* src = implicit_mode ? V : Y;
  const unsigned short* srch = reinterpret_cast<const unsigned short*>(V);

  for (long long row = (long long)blockIdx.x * 2 + wave; row < n_rows;
       row += (long long)gridDim.x * 2) {
    const long long start = indptr[row];
    const int n = (int)(indptr[row + 1] - start);
    if (n > NW || n <= NLO) continue;  // other variant / dense kernel
    if (n == 0 && NLO < 0) {
      for (int e = lane; e < F; e += 64)
        X[row * (long long)F + e] = 0.f;
      continue;
    }
    const bool probe = prof != nullptr && (row & 1023) == 0 && lane == 0;
    unsigned long long pt0 = 0, pt1 = 0, pt2 = 0, pt3 = 0;
    if (probe) pt0 = wall_clock64();

    // ---- stage factor rows + per-item weights ----
    // One coalesced load of the column ids first, THEN the factor-row
    // loads — with `indices[start+c]` inline each row load waited on its
    // own index fetch (PMC: 7166 wait vs 599 busy cycles per wave); via
    // LDS all n row loads issue back-to-back. The ids are staged through
    // the tv buffer (bit-cast; tv's real use starts after the last id
    // read). (Strided over e so F = 128 works with 64 lanes.)
    if (lane < n)
      tv[wave][lane] = __int_as_float(indices[start + lane]);
    wave_sync();
    for (int c = 0; c < n; ++c) {
      const long long col = __float_as_int(tv[wave][c]);
      if constexpr (BF16S) {
        const unsigned int* srow = reinterpret_cast<const unsigned int*>(
            srch + col * F);
        unsigned int* drow =
            reinterpret_cast<unsigned int*>(ylh + c * FPH);
        for (int e = lane; e < F / 2; e += 64) drow[e] = srow[e];
      } else {
        for (int e = lane; e < F; e += 64)
          yl[c * FP + e] = src[col * F + e];
      }
    }
    if (lane < n) {
      const float r = values[start + lane];
      if (implicit_mode) {
        // d = sqrt(alpha r) clamped: items with confidence ~0 still carry
        // their b-contribution (limit s_i -> c2_i as d -> 0)
        const float ar = alpha * r;
        const float d = sqrtf(ar > 1e-12f ? ar : 1e-12f);
        dv[wave][lane] = d;
        tv[wave][lane] = (1.f + ar) / d;     // rhs = c2 / d
      } else {
        tv[wave][lane] = r;                  // rhs = ratings
      }
    }
    wave_sync();
    if (probe) pt1 = wall_clock64();

    // ---- M = I + D G D  (implicit)  or  G + reg I  (explicit) ----
    // G is symmetric (G_ij = y_i^T B^-1 y_j): compute the upper triangle
    // only, mirror on write. Pair index advances incrementally — no
    // per-iteration integer division.
    const float reg = wr_scale ? lambda * (float)n : lambda;
    const int npairs = n * (n + 1) / 2;
    {
      // map lane → first (i, j) with j >= i in the flattened triangle
      int p = lane, i = 0;
      while (p >= n - i && i < n) { p -= n - i; ++i; }
      int j = i + p;
      for (int pp = lane; pp < npairs; pp += 64) {
        float dot;
        if constexpr (BF16S) {
          // b128 reads of 8 bf16; bf16->f32 is a 16-bit shift/mask on
          // the packed u32 (the bf16 bit pattern IS the f32 high half)
          const uint4* yi8 = reinterpret_cast<const uint4*>(&ylh[i * FPH]);
          const uint4* yj8 = reinterpret_cast<const uint4*>(&ylh[j * FPH]);
          f32x4_t acc4 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
          for (int q = 0; q < F / 8; ++q) {
            const uint4 a = yi8[q];
            const uint4 b = yj8[q];
            const unsigned int av[4] = {a.x, a.y, a.z, a.w};
            const unsigned int bv[4] = {b.x, b.y, b.z, b.w};
#pragma unroll
            for (int t = 0; t < 4; ++t) {
              acc4.x = fmaf(__uint_as_float(av[t] << 16),
                            __uint_as_float(bv[t] << 16), acc4.x);
              acc4.y = fmaf(__uint_as_float(av[t] & 0xffff0000u),
                            __uint_as_float(bv[t] & 0xffff0000u), acc4.y);
            }
          }
          dot = acc4.x + acc4.y;
        } else {
          const f32x4_t* yi = reinterpret_cast<const f32x4_t*>(&yl[i * FP]);
          const f32x4_t* yj = reinterpret_cast<const f32x4_t*>(&yl[j * FP]);
          f32x4_t acc4 = {0.f, 0.f, 0.f, 0.f};  // 2x v_pk_fma_f32 per q
#pragma unroll
          for (int q = 0; q < F / 4; ++q) acc4 += yi[q] * yj[q];
          dot = acc4.x + acc4.y + acc4.z + acc4.w;
        }
        if (implicit_mode) {
          dot *= dv[wave][i] * dv[wave][j];
          if (i == j) dot += 1.f;
        } else if (i == j) {
          dot += reg;
        }
        M[wave][i][j] = dot;
        M[wave][j][i] = dot;
        // advance 64 triangle slots
        j += 64;
        while (j >= n && i < n) { ++i; j -= n - i; }
      }
    }
    wave_sync();

    if (probe) pt2 = wall_clock64();
    // ---- Cholesky of M: register rows + quad pivot-column reads ----
    // lane = row of M in registers (mr); each finished column is written
    // into M's row k (M[k][j] = L[j][k]) and the trailing update reads it
    // back in b128 QUADS — the pure-LDS version compiled to one
    // ds_read_b32 + s_waitcnt per element (the same exposed-round-trip
    // disease fixed in the dense wave kernel, 1.7x there). A pure-shfl
    // variant without the quad reads measured slower (528 shfls); this
    // hybrid keeps the accumulators in registers and amortizes the LDS
    // latency 4-wide. Uniform `k < n` guards (no break) keep the k-loops
    // unrollable so mr[] stays in registers.
    float mr[NW];
#pragma unroll
    for (int q = 0; q < NW; ++q)
      mr[q] = (lane < n && q < n) ? M[wave][lane][q] : 0.f;
    float t = lane < n ? tv[wave][lane] : 0.f;
    wave_sync();
#pragma unroll
    for (int k = 0; k < NW; ++k) {
      if (k < n) {
        float mkk = __shfl(mr[k], k);
        mkk = mkk > 1e-30f ? mkk : 1e-30f;
        const float dinv = rsqrtf(mkk);
        const float ljk = lane > k ? mr[k] * dinv
                                   : (lane == k ? mkk * dinv : 0.f);
        mr[k] = ljk;
        if (lane < NW) M[wave][k][lane] = ljk;  // column k stored as row k
        wave_sync();
#pragma unroll
        for (int j4 = (k + 1) & ~3; j4 < NW; j4 += 4) {
          if (j4 < n) {
            const float4 q4 =
                *reinterpret_cast<const float4*>(&M[wave][k][j4]);
            if (lane > k) {
              // elements with j >= n update never-read mr slots (safe)
              if (j4 + 0 > k) mr[j4 + 0] = fmaf(-ljk, q4.x, mr[j4 + 0]);
              if (j4 + 1 > k) mr[j4 + 1] = fmaf(-ljk, q4.y, mr[j4 + 1]);
              if (j4 + 2 > k) mr[j4 + 2] = fmaf(-ljk, q4.z, mr[j4 + 2]);
              if (j4 + 3 > k) mr[j4 + 3] = fmaf(-ljk, q4.w, mr[j4 + 3]);
            }
          }
        }
        wave_sync();
      }
    }
    // forward solve L z = rhs (L[lane][k] = mr[k], registers)
#pragma unroll
    for (int k = 0; k < NW; ++k) {
      if (k < n) {
        const float lkk = __shfl(mr[k], k);
        const float zk = __shfl(t, k) / lkk;
        if (lane == k) t = zk;
        else if (lane > k) t = fmaf(-mr[k], zk, t);
      }
    }
    // back solve L^T s = z: L[k][lane] = M[lane][k] (row `lane` holds
    // column `lane`)
    wave_sync();
#pragma unroll
    for (int k = NW - 1; k >= 0; --k) {
      if (k < n) {
        const float lkk = __shfl(mr[k], k);
        const float xk = __shfl(t, k) / lkk;
        if (lane == k) t = xk;
        else if (lane < k) t = fmaf(-M[wave][lane][k], xk, t);
      }
    }
    if (implicit_mode) t *= dv[wave][lane < n ? lane : 0];  // s = d t
    if (lane < n) tv[wave][lane] = t;
    wave_sync();

    if (probe) pt3 = wall_clock64();
    // ---- emit sum_i s_i v_i (implicit: z, host solves X = Z L^-1)
    //      or   sum_i s_i y_i (explicit: x directly) ----
    for (int e = lane; e < F; e += 64) {
      float x = 0.f;
      for (int c = 0; c < n; ++c) {
        float ye;
        if constexpr (BF16S)
          ye = __uint_as_float((unsigned int)ylh[c * FPH + e] << 16);
        else
          ye = yl[c * FP + e];
        x = fmaf(tv[wave][c], ye, x);
      }
      X[row * (long long)F + e] = x;
    }
    if (probe) {
      const unsigned long long pt4 = wall_clock64();
      atomicAdd(&prof[0], pt1 - pt0);   // stage
      atomicAdd(&prof[1], pt2 - pt1);   // G build
      atomicAdd(&prof[2], pt3 - pt2);   // M solve
      atomicAdd(&prof[3], pt4 - pt3);   // emit
      atomicAdd(&prof[4], 1ull);        // samples
    }
    wave_sync();
  }
}

// ---------------------------------------------------------------------------
// Dual-row Woodbury solver (round 2): TWO rows per wave, one per 32-lane
// half. The phase probe showed the single-row kernel issue-bound with the
// register Cholesky M-solve at 66% of time while only n <= 32 of 64 lanes
// were active — packing a second row into the idle half halves the
// per-row issue count of the solve (both halves factor concurrently in
// one instruction stream; __shfl with a computed (lane & 32) + k source
// stays within each half). Staging/G/emit keep full-wave cooperation per
// row, so their per-row cost is unchanged.
// ---------------------------------------------------------------------------

template <int F, int NW, int NLO, bool BF16S = false>
__global__ __launch_bounds__(128, 3) void als_woodbury2_kernel(
    const long long* __restrict__ indptr,
    const int* __restrict__ indices,
    const float* __restrict__ values,
    const float* __restrict__ Y,
    const float* __restrict__ V,
    float* __restrict__ X,
    int n_rows,
    float lambda,
    float alpha,
    int implicit_mode,
    int wr_scale,
    unsigned long long* prof)
{
  constexpr int FP = F + 4;
  constexpr int FPH = F + 8;  // bf16 row stride (ushorts)
  constexpr int MP = NW + 4;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int h = lane >> 5;    // wave half = which of the two rows
  const int r = lane & 31;    // sub-lane within the half

  __shared__ float Yl[2][2][NW][BF16S ? (FPH / 2) : FP];  // [wave][half]
  constexpr int YROW = BF16S ? (FPH / 2) : FP;  // floats per staged row
  __shared__ float M[2][2][NW][MP];
  __shared__ float tv[2][2][NW];
  __shared__ float dv[2][2][NW];

  const float* src = implicit_mode ? V : Y;
  const unsigned short* srch = reinterpret_cast<const unsigned short*>(V);

  for (long long base = ((long long)blockIdx.x * 2 + wave) * 2;
       base < n_rows; base += (long long)gridDim.x * 4) {
    // ---- per-half row bookkeeping (uniform across the wave) ----
    int nh[2];
    long long starth[2];
    bool act[2];
#pragma unroll
    for (int hh = 0; hh < 2; ++hh) {
      const long long row = base + hh;
      if (row < n_rows) {
        starth[hh] = indptr[row];
        nh[hh] = (int)(indptr[row + 1] - starth[hh]);
        act[hh] = nh[hh] <= NW && nh[hh] > NLO;
        if (nh[hh] == 0 && NLO < 0) {
          for (int e = lane; e < F; e += 64)
            X[row * (long long)F + e] = 0.f;
          act[hh] = false;
        }
      } else {
        act[hh] = false;
        nh[hh] = 0;
        starth[hh] = 0;
      }
    }
    if (!act[0] && !act[1]) continue;
    const bool probe = prof != nullptr && (base & 1023) == 0 && lane == 0
                       && act[0];
    unsigned long long pt0 = 0, pt1 = 0, pt2 = 0, pt3 = 0;
    if (probe) pt0 = wall_clock64();

    // ---- stage both rows (full-wave per row, same as single-row) ----
#pragma unroll
    for (int hh = 0; hh < 2; ++hh) {
      if (!act[hh]) continue;
      const int n = nh[hh];
      const long long start = starth[hh];
      float* yl = &Yl[wave][hh][0][0];
      unsigned short* ylh = reinterpret_cast<unsigned short*>(yl);
      if (lane < n)
        tv[wave][hh][lane] = __int_as_float(indices[start + lane]);
      wave_sync();
      for (int c = 0; c < n; ++c) {
        const long long col = __float_as_int(tv[wave][hh][c]);
        if constexpr (BF16S) {
          const unsigned int* srow = reinterpret_cast<const unsigned int*>(
              srch + col * F);
          unsigned int* drow =
              reinterpret_cast<unsigned int*>(ylh + c * FPH);
          for (int e = lane; e < F / 2; e += 64) drow[e] = srow[e];
        } else {
          for (int e = lane; e < F; e += 64)
            yl[c * FP + e] = src[col * F + e];
        }
      }
      if (lane < n) {
        const float v = values[start + lane];
        if (implicit_mode) {
          const float ar = alpha * v;
          const float d = sqrtf(ar > 1e-12f ? ar : 1e-12f);
          dv[wave][hh][lane] = d;
          tv[wave][hh][lane] = (1.f + ar) / d;
        } else {
          tv[wave][hh][lane] = v;
        }
      }
    }
    wave_sync();
    if (probe) pt1 = wall_clock64();

    // ---- G build: each half computes its own row's pair triangle ----
    const int n = nh[h];           // this half's row size (per-lane)
    {
      const float* yl = &Yl[wave][h][0][0];
      const unsigned short* ylh =
          reinterpret_cast<const unsigned short*>(yl);
      const float reg = wr_scale ? lambda * (float)n : lambda;
      const int npairs = act[h] ? n * (n + 1) / 2 : 0;
      int p = r, i = 0;
      while (p >= n - i && i < n) { p -= n - i; ++i; }
      int j = i + p;
      for (int pp = r; pp < npairs; pp += 32) {
        float dot;
        if constexpr (BF16S) {
          const uint4* yi8 = reinterpret_cast<const uint4*>(&ylh[i * FPH]);
          const uint4* yj8 = reinterpret_cast<const uint4*>(&ylh[j * FPH]);
          f32x4_t acc4 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
          for (int q = 0; q < F / 8; ++q) {
            const uint4 a = yi8[q];
            const uint4 b = yj8[q];
            const unsigned int av[4] = {a.x, a.y, a.z, a.w};
            const unsigned int bv[4] = {b.x, b.y, b.z, b.w};
#pragma unroll
            for (int t = 0; t < 4; ++t) {
              acc4.x = fmaf(__uint_as_float(av[t] << 16),
                            __uint_as_float(bv[t] << 16), acc4.x);
              acc4.y = fmaf(__uint_as_float(av[t] & 0xffff0000u),
                            __uint_as_float(bv[t] & 0xffff0000u), acc4.y);
            }
          }
          dot = acc4.x + acc4.y;
        } else {
          const f32x4_t* yi = reinterpret_cast<const f32x4_t*>(&yl[i * FP]);
          const f32x4_t* yj = reinterpret_cast<const f32x4_t*>(&yl[j * FP]);
          f32x4_t acc4 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
          for (int q = 0; q < F / 4; ++q) acc4 += yi[q] * yj[q];
          dot = acc4.x + acc4.y + acc4.z + acc4.w;
        }
        if (implicit_mode) {
          dot *= dv[wave][h][i] * dv[wave][h][j];
          if (i == j) dot += 1.f;
        } else if (i == j) {
          dot += reg;
        }
        M[wave][h][i][j] = dot;
        M[wave][h][j][i] = dot;
        j += 32;
        while (j >= n && i < n) { ++i; j -= n - i; }
      }
    }
    wave_sync();
    if (probe) pt2 = wall_clock64();

    // ---- Cholesky: both halves factor their M concurrently ----
    float mr[NW];
#pragma unroll
    for (int q = 0; q < NW; ++q)
      mr[q] = (r < n && q < n) ? M[wave][h][r][q] : 0.f;
    float t = r < n ? tv[wave][h][r] : 0.f;
    wave_sync();
    const int hb = lane & 32;     // shfl base of this half
#pragma unroll
    for (int k = 0; k < NW; ++k) {
      if (k < n) {                // per-lane guard (halves may differ)
        float mkk = __shfl(mr[k], hb + k);
        mkk = mkk > 1e-30f ? mkk : 1e-30f;
        const float dinv = rsqrtf(mkk);
        const float ljk = r > k ? mr[k] * dinv
                                : (r == k ? mkk * dinv : 0.f);
        mr[k] = ljk;
        if (r < NW) M[wave][h][k][r] = ljk;  // column k stored as row k
      }
      wave_sync();
      if (k < n) {
#pragma unroll
        for (int j4 = (k + 1) & ~3; j4 < NW; j4 += 4) {
          if (j4 < n) {
            const float4 q4 =
                *reinterpret_cast<const float4*>(&M[wave][h][k][j4]);
            if (r > k) {
              if (j4 + 0 > k) mr[j4 + 0] = fmaf(-mr[k], q4.x, mr[j4 + 0]);
              if (j4 + 1 > k) mr[j4 + 1] = fmaf(-mr[k], q4.y, mr[j4 + 1]);
              if (j4 + 2 > k) mr[j4 + 2] = fmaf(-mr[k], q4.z, mr[j4 + 2]);
              if (j4 + 3 > k) mr[j4 + 3] = fmaf(-mr[k], q4.w, mr[j4 + 3]);
            }
          }
        }
      }
      wave_sync();
    }
    // forward solve L z = rhs
#pragma unroll
    for (int k = 0; k < NW; ++k) {
      if (k < n) {
        const float lkk = __shfl(mr[k], hb + k);
        const float zk = __shfl(t, hb + k) / lkk;
        if (r == k) t = zk;
        else if (r > k) t = fmaf(-mr[k], zk, t);
      }
    }
    // back solve L^T s = z
    wave_sync();
#pragma unroll
    for (int k = NW - 1; k >= 0; --k) {
      if (k < n) {
        const float lkk = __shfl(mr[k], hb + k);
        const float xk = __shfl(t, hb + k) / lkk;
        if (r == k) t = xk;
        else if (r < k) t = fmaf(-M[wave][h][r][k], xk, t);
      }
    }
    if (implicit_mode) t *= dv[wave][h][r < n ? r : 0];
    if (r < n && act[h]) tv[wave][h][r] = t;
    wave_sync();
    if (probe) pt3 = wall_clock64();

    // ---- emit both rows (full-wave per row) ----
#pragma unroll
    for (int hh = 0; hh < 2; ++hh) {
      if (!act[hh]) continue;
      const float* yl = &Yl[wave][hh][0][0];
      const unsigned short* ylh =
          reinterpret_cast<const unsigned short*>(yl);
      const int nn = nh[hh];
      const long long row = base + hh;
      for (int e = lane; e < F; e += 64) {
        float x = 0.f;
        for (int c = 0; c < nn; ++c) {
          float ye;
          if constexpr (BF16S)
            ye = __uint_as_float(
                (unsigned int)ylh[c * FPH + e] << 16);
          else
            ye = yl[c * FP + e];
          x = fmaf(tv[wave][hh][c], ye, x);
        }
        X[row * (long long)F + e] = x;
      }
    }
    if (probe) {
      const unsigned long long pt4 = wall_clock64();
      atomicAdd(&prof[0], pt1 - pt0);
      atomicAdd(&prof[1], pt2 - pt1);
      atomicAdd(&prof[2], pt3 - pt2);
      atomicAdd(&prof[3], pt4 - pt3);
      atomicAdd(&prof[4], 1ull);
    }
    wave_sync();
  }
}

// Variant flag for als_solve_kernel: skip rows the Woodbury kernel owns.

// ---------------------------------------------------------------------------
// launcher
// ---------------------------------------------------------------------------

// instantiation guard: the MFMA-Gramian wave kernel exists only at F=64
template <int FF>
static void launch_wave_bf16g(int grid_w, hipStream_t stream,
    const long long* indptr, const int* indices, const float* values,
    const float* Y, const float* YtY, float* X, int n_rows, float lambda,
    float alpha, int implicit_mode, int wr_scale, int skip, int wave_hi) {
  if constexpr (FF == 64) {
    hipLaunchKernelGGL((als_solve_wave_kernel<64, true>), dim3(grid_w),
                       dim3(128), 0, stream, indptr, indices, values, Y,
                       YtY, X, n_rows, lambda, alpha, implicit_mode,
                       wr_scale, skip, wave_hi);
  }
}

extern "C" void launch_als_solve(
    const long long* indptr, const int* indices, const float* values,
    const float* Y, const float* YtY, const float* V, float* X,
    int n_rows, int f, float lambda, float alpha,
    int implicit_mode, int wr_scale, int which,
    unsigned long long* prof, hipStream_t stream)
{
  // >> 256 workgroups to fill 256 CUs across 8 XCDs; grid-stride for huge
  // row counts.
  // which: 0 = both passes (explicit mode — X is direct either way),
  //        1 = Woodbury only (implicit: emits Z for the host-side trsm),
  //        2 = dense only, skipping Woodbury-owned rows (implicit: fills
  //            the big rows with direct X after the trsm).
  if (n_rows <= 0) return;
  // wave/woodbury kernels: 2 rows per 128-thread workgroup
  long long wg = ((long long)n_rows + 1) / 2;
  int grid_w = (int)(wg < (1 << 20) ? wg : (1 << 20));
  int grid_b = n_rows < (1 << 20) ? n_rows : (1 << 20);
  const bool woodbury = which != 2 &&
      (implicit_mode ? V != nullptr : true);
  const int skip = (woodbury || which == 2) ? WOODBURY_MAX_NNZ : -1;
  const bool dense = which != 1;
  // PIO_ALS_DENSE_SPLIT=N routes rows with nnz > N to the
  // workgroup-per-row kernel (256 threads/row — 4x the per-row
  // parallelism of the wave kernel for very dense rows, e.g. the item
  // side at 8 GPUs where nnz ~ 200). 0/unset = wave kernel takes all.
  const char* e_split = getenv("PIO_ALS_DENSE_SPLIT");
  const int dense_split = e_split ? atoi(e_split) : 0;
  const int wave_hi = dense_split > 0 ? dense_split : 0x7fffffff;
#define LAUNCH_WAVE(FF)                                                      \
  if (dense && use_bf16s_g && FF == 64) {                                    \
  launch_wave_bf16g<FF>(grid_w, stream, indptr, indices, values, Y, YtY, X, \
                        n_rows, lambda, alpha, implicit_mode, wr_scale,      \
                        skip, wave_hi);                                      \
  if (dense_split > 0)                                                       \
    hipLaunchKernelGGL((als_solve_kernel<FF>), dim3(grid_b), dim3(256), 0,   \
                       stream, indptr, indices, values, Y, YtY, X, n_rows,   \
                       lambda, alpha, implicit_mode, wr_scale, dense_split); \
  } else if (dense) {                                                        \
  hipLaunchKernelGGL((als_solve_wave_kernel<FF>), dim3(grid_w), dim3(128),   \
                     0, stream, indptr, indices, values, Y, YtY, X, n_rows,  \
                     lambda, alpha, implicit_mode, wr_scale, skip, wave_hi); \
  if (dense_split > 0)                                                       \
    hipLaunchKernelGGL((als_solve_kernel<FF>), dim3(grid_b), dim3(256), 0,   \
                       stream, indptr, indices, values, Y, YtY, X, n_rows,   \
                       lambda, alpha, implicit_mode, wr_scale, dense_split); \
  }
#define LAUNCH_BLOCK(FF)                                                     \
  if (dense)                                                                 \
  hipLaunchKernelGGL((als_solve_kernel<FF>), dim3(grid_b), dim3(256), 0,     \
                     stream, indptr, indices, values, Y, YtY, X, n_rows,     \
                     lambda, alpha, implicit_mode, wr_scale, skip)
  // PIO_ALS_DUAL=1 routes the Woodbury rows through the dual-row kernel
  // (two rows per wave) — A/B lever for the round-2 M-solve rework
  const char* e_dual = getenv("PIO_ALS_DUAL");
  const bool use_dual = e_dual != nullptr && e_dual[0] == '1';
  // PIO_ALS_STAGE_BF16=1: V arrives as bf16 (the Python side casts) and
  // the Woodbury kernels stage it as bf16 — implicit mode only
  const char* e_bf16 = getenv("PIO_ALS_STAGE_BF16");
  const bool stage_bf16 = e_bf16 != nullptr && e_bf16[0] == '1';
  const bool use_bf16s = stage_bf16 && implicit_mode && V != nullptr;
  // the MFMA-Gramian dense variant has its OWN opt-in: it measured
  // SLOWER at bench shapes (the wave kernel is gather-latency-bound,
  // so the VALU Gramian it replaces was hidden anyway) and must not
  // ride along with the beneficial PIO_ALS_STAGE_BF16 combo
  const char* e_mg = getenv("PIO_ALS_MFMA_GRAMIAN");
  const bool use_bf16s_g = e_mg != nullptr && e_mg[0] == '1';
  long long wg4 = ((long long)n_rows + 3) / 4;
  int grid_w2 = (int)(wg4 < (1 << 20) ? wg4 : (1 << 20));
#define LAUNCH_WOODBURY(FF)                                                  \
  if (woodbury && use_dual && use_bf16s) {                                   \
    hipLaunchKernelGGL((als_woodbury2_kernel<FF, 20, -1, true>),             \
                       dim3(grid_w2), dim3(128), 0, stream, indptr,          \
                       indices, values, Y, V, X, n_rows, lambda, alpha,      \
                       implicit_mode, wr_scale, prof);                       \
    hipLaunchKernelGGL((als_woodbury2_kernel<FF, 24, 20, true>),             \
                       dim3(grid_w2), dim3(128), 0, stream, indptr,          \
                       indices, values, Y, V, X, n_rows, lambda, alpha,      \
                       implicit_mode, wr_scale, prof);                       \
    hipLaunchKernelGGL((als_woodbury_kernel<FF, 32, 24, true>),              \
                       dim3(grid_w), dim3(128), 0, stream, indptr, indices,  \
                       values, Y, V, X, n_rows, lambda, alpha,               \
                       implicit_mode, wr_scale, prof);                       \
  } else if (woodbury && use_dual) {                                         \
    /* dual for nnz<=24; the NW=32 dual instantiation spills (mr[32] x   */  \
    /* dual-row bookkeeping exceeds the register budget) so rows 25-32   */  \
    /* keep the single-row kernel                                        */  \
    hipLaunchKernelGGL((als_woodbury2_kernel<FF, 20, -1>), dim3(grid_w2),    \
                       dim3(128), 0, stream, indptr, indices, values, Y, V,  \
                       X, n_rows, lambda, alpha, implicit_mode, wr_scale,    \
                       prof);                                                \
    hipLaunchKernelGGL((als_woodbury2_kernel<FF, 24, 20>), dim3(grid_w2),    \
                       dim3(128), 0, stream, indptr, indices, values, Y, V,  \
                       X, n_rows, lambda, alpha, implicit_mode, wr_scale,    \
                       prof);                                                \
    hipLaunchKernelGGL((als_woodbury_kernel<FF, 32, 24>), dim3(grid_w),      \
                       dim3(128), 0, stream, indptr, indices, values, Y, V,  \
                       X, n_rows, lambda, alpha, implicit_mode, wr_scale,    \
                       prof);                                                \
  } else if (woodbury && use_bf16s) {                                        \
    hipLaunchKernelGGL((als_woodbury_kernel<FF, 20, -1, true>),              \
                       dim3(grid_w), dim3(128), 0, stream, indptr, indices,  \
                       values, Y, V, X, n_rows, lambda, alpha,               \
                       implicit_mode, wr_scale, prof);                       \
    hipLaunchKernelGGL((als_woodbury_kernel<FF, 24, 20, true>),              \
                       dim3(grid_w), dim3(128), 0, stream, indptr, indices,  \
                       values, Y, V, X, n_rows, lambda, alpha,               \
                       implicit_mode, wr_scale, prof);                       \
    hipLaunchKernelGGL((als_woodbury_kernel<FF, 32, 24, true>),              \
                       dim3(grid_w), dim3(128), 0, stream, indptr, indices,  \
                       values, Y, V, X, n_rows, lambda, alpha,               \
                       implicit_mode, wr_scale, prof);                       \
  } else if (woodbury) {                                                     \
    hipLaunchKernelGGL((als_woodbury_kernel<FF, 20, -1>), dim3(grid_w),      \
                       dim3(128), 0, stream, indptr, indices, values, Y, V,  \
                       X, n_rows, lambda, alpha, implicit_mode, wr_scale,    \
                       prof);                                                \
    hipLaunchKernelGGL((als_woodbury_kernel<FF, 24, 20>), dim3(grid_w),      \
                       dim3(128), 0, stream, indptr, indices, values, Y, V,  \
                       X, n_rows, lambda, alpha, implicit_mode, wr_scale,    \
                       prof);                                                \
    hipLaunchKernelGGL((als_woodbury_kernel<FF, 32, 24>), dim3(grid_w),      \
                       dim3(128), 0, stream, indptr, indices, values, Y, V,  \
                       X, n_rows, lambda, alpha, implicit_mode, wr_scale,    \
                       prof);                                                \
  }
  switch (f) {
    case 16: LAUNCH_WOODBURY(16); LAUNCH_WAVE(16); break;
    case 32: LAUNCH_WOODBURY(32); LAUNCH_WAVE(32); break;
    case 64: LAUNCH_WOODBURY(64); LAUNCH_WAVE(64); break;
    case 128: LAUNCH_WOODBURY(128); LAUNCH_BLOCK(128); break;
    default: break;  // caller validates
  }
#undef LAUNCH_WAVE
#undef LAUNCH_BLOCK
#undef LAUNCH_WOODBURY
}
