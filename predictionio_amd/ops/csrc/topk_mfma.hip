// MFMA-based masked top-K scoring for MI355X (gfx950, CDNA4).
//
// v5 of the serve hot path: the B x N x F score computation runs on the
// matrix cores (v_mfma_f32_16x16x32_bf16, bf16 inputs / fp32 accumulate)
// instead of the VALU fp32 dot loop of topk_score_kernel (v3). Replaces
// the same reference semantics (recommendProductsWithFilter
// examples/.../ALSModel.scala:44-60, ecommerce predictKnownUser
// ECommAlgorithm.scala:471-506, similarproduct cosine top-N
// ALSAlgorithm.scala:168-242); the Python wrapper re-checks the bf16
// survivors against the fp32 factors so served scores stay exact.
//
// Geometry (per 256-thread workgroup, 4 waves):
//  - the block owns 64 queries; wave w owns queries [16w, 16w+16)
//  - item slices: grid = (ublocks, n_slices); blockIdx.x is the ublock
//    so concurrently-resident workgroups share one Y slice through L3
//  - per chunk of TM_CHUNK=64 items: Y rows are reg-staged into LDS as
//    bf16 with an XOR swizzle ((row & SWM) << 4 on the byte offset, the
//    T2 bank-conflict fix for "different rows, same col-range" b128
//    reads), then each wave runs (TM_CHUNK/16) x (F/32) MFMAs:
//       A = Y tile   (lane holds Y[item = l&15][k = (l>>4)*8 + i])
//       B = X^T tile (lane holds X[query = l&15][k = (l>>4)*8 + i]),
//    accumulating D[item][query] in fp32 (D: col = l&15 = query,
//    row = (l>>4)*4 + reg = item) — operand maps per the CK xdlops
//    contract and the measured gfx950 C/D layout.
//
// Top-K epilogue (v2 — the round-2 phase probe showed the v1 per-lane
// list scheme spending 63% of kernel time on insert machinery):
//  - ONE top-K list per (query, workgroup) in LDS, with the running
//    K-th-best threshold also in LDS (th_lds[wave*16 + query]).
//  - common case per chunk: each lane max-reduces its 16 accumulator
//    values (15 v_max), reads the shared threshold (one broadcast
//    ds_read) and skips everything else — ~25 instructions per
//    64 items x 16 queries.
//  - rare case (some lane's max beats the threshold): the 4 lanes of a
//    query's quad take turns (serialized by lane group, exec-masked —
//    a single wave executes groups in program order, and within one
//    group every active lane owns a DIFFERENT query, so list writes
//    never contend) scanning their values and inserting; masks/bans are
//    only consulted here.
//  - a shared per-query threshold is strictly tighter than v1's
//    per-lane-group thresholds, cutting total inserts ~4x (inserts per
//    query = K ln(items/K) per LIST, and lists per query dropped 4x),
//    and the list LDS shrinks 4x (64 lists instead of 256), lifting
//    occupancy from 3 to 7 workgroups/CU.
//
// The X fragments live in registers for the whole kernel (loaded once);
// item staging is software-pipelined through registers, with the next
// chunk's loads issued AFTER the second barrier so they fly under the
// MFMA phase (a __syncthreads compiles to s_waitcnt vmcnt(0) — loads
// issued before it would be drained at the barrier).

#include <float.h>
#include <hip/hip_runtime.h>

#define TM_CHUNK 64   // items staged per LDS pass
#define TM_WAVES 4
#define TM_QPW 16     // queries per wave
#define TM_UPB (TM_WAVES * TM_QPW)  // queries per block

typedef __attribute__((ext_vector_type(8))) unsigned short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;

// Order-preserving float<->u32 map for atomicMax on scores (classic
// sign-flip transform: negative floats map to ~bits, non-negative to
// bits|0x80000000, making unsigned order == float order).
__device__ __forceinline__ unsigned tm_enc(float f) {
  unsigned u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}
__device__ __forceinline__ float tm_dec(unsigned k) {
  return __uint_as_float((k & 0x80000000u) ? (k & 0x7FFFFFFFu) : ~k);
}

__device__ __forceinline__ bool tm_in_sorted(const int* arr, int n, int x) {
  int lo = 0, hi = n - 1;
  while (lo <= hi) {
    int mid = (lo + hi) >> 1;
    int v = arr[mid];
    if (v == x) return true;
    if (v < x) lo = mid + 1; else hi = mid - 1;
  }
  return false;
}

// QB = query blocks per workgroup (each 64 queries, scored SEQUENTIALLY
// per staged chunk). QB=2 amortizes the staging phase (the round-2 probe
// measured ~50/50 stage/score) over 2x the MFMA work and HALVES the
// total HBM item traffic (half as many ublocks read the full Y set);
// the cost is 2x list LDS (occupancy ~7 -> ~5 WGs/CU) and 2x epilogue.
// QB=2 needs ~2x the live registers (both query blocks' X fragments +
// insert state): launch_bounds min-waves 6 would cap it at 80 VGPRs and
// spill the hot loop to scratch (measured 3-4x slower — scratch counts
// toward vmcnt and drains the staging pipeline every chunk), so QB=2
// instantiations get min-waves 4 (128 VGPRs; LDS caps occupancy at
// ~5 WGs/CU for K=20 anyway).
template <int F, bool PROF, bool DBUF, int CH = TM_CHUNK,
          int NWAVES = TM_WAVES, bool GLL = false, int QB = 1>
__global__ __launch_bounds__(NWAVES * 64, QB == 1 ? 24 / NWAVES : 4)
void topk_mfma_kernel(
    const unsigned short* __restrict__ Xq,   // B x F bf16
    const unsigned short* __restrict__ Y,    // N x F bf16
    const uint8_t* __restrict__ item_mask,   // N or nullptr
    const long long* __restrict__ ban_indptr,
    const int* __restrict__ ban_indices,
    float* __restrict__ out_val,             // B x n_slices x K
    int* __restrict__ out_idx,
    int B, long long N, int K, int n_slices, int item_base,
    unsigned long long* prof,
    // optional [B] cross-slice threshold (tm_enc-coded, init 0): the
    // concurrently-running WGs of a query's OTHER slices publish their
    // K-th-best through L2, so each slice stops re-paying the full
    // insert ramp from -inf (insert volume was ~linear in n_slices)
    unsigned* __restrict__ th_g,
    int kflags)  // bit 0: disable the ban bloom (A/B hook)
{
  constexpr int ROWB = F * 2;            // bytes per staged Y row
  constexpr int SWM = (F >= 64) ? 7 : 3; // XOR-swizzle row mask
  constexpr int KS = F / 32;             // MFMA K-steps per dot product
  constexpr int IFR = CH / 16;           // item fragments per chunk
  extern __shared__ char lds_raw[];
  // DBUF: two ys buffers, ONE barrier per chunk (compute buf[i&1] while
  // draining the next chunk into buf[(i+1)&1])
  unsigned short* ys = reinterpret_cast<unsigned short*>(lds_raw);
  float* topv = reinterpret_cast<float*>(
      lds_raw + (DBUF ? 2 : 1) * CH * ROWB);
  const int KP = K + 1;  // stride coprime with the 32 banks (v3 lesson)
  constexpr int BS = NWAVES * 64;        // block size
  constexpr int UPB = NWAVES * TM_QPW;   // queries per query block
  constexpr int UPBT = UPB * QB;         // queries per workgroup
  int* topi = reinterpret_cast<int*>(topv + UPBT * KP);
  float* th_lds = reinterpret_cast<float*>(topi + UPBT * KP);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int lg = lane >> 4;   // lane group: k-slice on input, item rows out
  const int lq = lane & 15;   // query column (B operand / D col)

  const bool probe = PROF && tid == 0;
  unsigned long long pt = 0, acc_setup = 0, acc_stage = 0, acc_score = 0;
  if (probe) pt = wall_clock64();

  const long long u0 = (long long)blockIdx.x * UPBT;
  long long guser[QB];
  bool has_user[QB];
#pragma unroll
  for (int qb = 0; qb < QB; ++qb) {
    guser[qb] = u0 + qb * UPB + wave * TM_QPW + lq;
    has_user[qb] = guser[qb] < B;
  }
  const int slice = blockIdx.y;
  const long long per = (N + n_slices - 1) / n_slices;
  const long long it0 = (long long)slice * per;
  const long long it1 = min(N, it0 + per);

  // ---- X fragments: one b128 per K-step, held for the whole kernel
  bf16x8 xf[QB][KS];
#pragma unroll
  for (int qb = 0; qb < QB; ++qb)
#pragma unroll
    for (int ks = 0; ks < KS; ++ks) {
      if (has_user[qb]) {
        xf[qb][ks] = *reinterpret_cast<const bf16x8*>(
            &Xq[guser[qb] * F + ks * 32 + lg * 8]);
      } else {
        bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        xf[qb][ks] = z;
      }
    }

  // ---- init the per-query top-K lists + shared thresholds
  for (int e = tid; e < UPBT * KP; e += BS) {
    topv[e] = -FLT_MAX;
    topi[e] = -1;
  }
  if (tid < UPBT) th_lds[tid] = -FLT_MAX;
  __syncthreads();

  const int* ban[QB];
  int bn[QB];
  // per-query 64-bit bloom over the ban list: the insert path's binary
  // search is ~5 DEPENDENT global loads inside the serialized
  // lane-group loop, so a one-register membership pre-test that skips
  // it for definitely-unbanned items pays for itself after a handful
  // of inserts (~47% false-positive at 30 bans; saturates harmlessly
  // for huge lists). Built once at setup from this lane's own list.
  unsigned long long bloom[QB];
#pragma unroll
  for (int qb = 0; qb < QB; ++qb) {
    ban[qb] = nullptr;
    bn[qb] = 0;
    bloom[qb] = 0ull;
    if (ban_indptr != nullptr && has_user[qb]) {
      const long long b0 = ban_indptr[guser[qb]];
      bn[qb] = (int)(ban_indptr[guser[qb] + 1] - b0);
      ban[qb] = ban_indices + b0;
      if ((kflags & 1) || bn[qb] > 512) {
        // disabled, or the list is big enough that the 64-bit bloom
        // would saturate anyway — skip the build, every check falls
        // through to the binary search (measured 0.998x at 300 bans)
        bloom[qb] = ~0ull;
      } else {
        for (int t = 0; t < bn[qb]; ++t)
          bloom[qb] |= 1ull
              << ((unsigned)((unsigned)ban[qb][t] * 2654435761u) >> 26);
      }
    }
  }
  if (probe) {
    const unsigned long long now = wall_clock64();
    acc_setup = now - pt;
    pt = now;
  }

  // ---- software-pipelined staging registers (granules of 16 B)
  constexpr int NG = (CH * ROWB) / 16 / BS;         // granules per thread
  static_assert(NG >= 1, "chunk must cover one granule per thread");
  u32x4 stg[NG];
  // per-thread element offsets are chunk-invariant: loads use ONE
  // running base pointer advanced by a constant per chunk (the naive
  // form re-did ~20 64-bit address ops per chunk in the hot loop)
  int goff[NG];
  int grow[NG];
#pragma unroll
  for (int r = 0; r < NG; ++r) {
    const int lin = (tid + r * BS) * 16;
    grow[r] = lin / ROWB;
    goff[r] = grow[r] * F + (lin % ROWB) / 2;
  }
  auto load_stg = [&](long long cbase) {
    const unsigned short* yb_g = Y + cbase * F;
    const bool tail = cbase + CH > it1;
    if (!tail) {
#pragma unroll
      for (int r = 0; r < NG; ++r)
        stg[r] = *reinterpret_cast<const u32x4*>(&yb_g[goff[r]]);
    } else {
#pragma unroll
      for (int r = 0; r < NG; ++r)
        stg[r] = (cbase + grow[r] < it1)
            ? *reinterpret_cast<const u32x4*>(&yb_g[goff[r]])
            : u32x4{0u, 0u, 0u, 0u};
    }
  };
  auto drain_to = [&](unsigned short* yb) {
#pragma unroll
    for (int r = 0; r < NG; ++r) {
      const int lin = (tid + r * BS) * 16;
      const int row = lin / ROWB;
      const int col = lin % ROWB;
      const int dst = row * ROWB + (col ^ ((row & SWM) << 4));
      *reinterpret_cast<u32x4*>(reinterpret_cast<char*>(yb) + dst) = stg[r];
    }
  };
  auto ybuf = [&](int i) -> unsigned short* {
    return ys + (DBUF ? (size_t)(i & 1) * CH * F : 0);
  };
  // GLL: async global->LDS DMA (no staging registers, no drain
  // ds_writes). The LDS destination is wave-uniform base + lane*16
  // (linear), so the XOR swizzle moves to the per-lane GLOBAL source
  // address — the involution makes the LDS image identical to a
  // swizzled write (guide rule 21). Tail rows clamp the source row;
  // their stale granules are filtered by the epilogue's li < lim.
  auto issue_gll = [&](long long cbase) {
#pragma unroll
    for (int r = 0; r < NG; ++r) {
      const int g = tid + r * BS;
      const int lin = g * 16;
      const int row = lin / ROWB;
      const int col = lin % ROWB;
      long long grow_g = cbase + row;
      if (grow_g >= N) grow_g = N - 1;
      const unsigned short* src =
          Y + grow_g * F + ((col ^ ((row & SWM) << 4)) >> 1);
      unsigned short* dst =
          ys + ((size_t)(tid & ~63) + (size_t)r * BS) * 8;  // wave base
      __builtin_amdgcn_global_load_lds(
          reinterpret_cast<const unsigned int*>(src),
          reinterpret_cast<unsigned int*>(dst), 16, 0, 0);
    }
  };
  // pipelined cross-slice threshold: the register holds the value read
  // one fold period ago (tm_enc-coded; 0 decodes below every real
  // score, so the first fold is a no-op)
  unsigned gth_pipe[QB];
#pragma unroll
  for (int qb = 0; qb < QB; ++qb)
    gth_pipe[qb] = (th_g != nullptr && lg == 0 && has_user[qb])
        ? th_g[guser[qb]] : 0u;
  const int n_chunks = (int)((it1 - it0 + CH - 1) / CH);
  if (!GLL) load_stg(it0);
  if (DBUF) {
    // prologue fill of buffer 0; chunk 1's loads fly under chunk 0
    drain_to(ybuf(0));
    __syncthreads();
    if (n_chunks > 1) load_stg(it0 + CH);
  }

  for (int ci = 0; ci < n_chunks; ++ci) {
    const long long base = it0 + (long long)ci * CH;
    unsigned short* yb = ybuf(ci);
    if (GLL) {
      __syncthreads();  // all waves done reading the previous chunk
      issue_gll(base);
      __syncthreads();  // drains the in-flight LDS DMA (vmcnt)
    } else if (!DBUF) {
      __syncthreads();  // all waves done reading the previous chunk
      drain_to(yb);
      // fold the cross-slice global threshold into the local one every
      // 8 chunks; each wave's lg==0 lanes update only their own wave's
      // lists, and the barrier below orders the write before the
      // epilogue's reads. PIPELINED: consume the value LOADED 8 CHUNKS
      // AGO (a barrier since then already paid its vmcnt, so the fold
      // never stalls on the load) and issue the next one; thresholds
      // are monotonic so an 8-chunk-stale value only prunes less.
      if (th_g != nullptr && (ci & 7) == 0 && lg == 0) {
#pragma unroll
        for (int qb = 0; qb < QB; ++qb) {
          if (has_user[qb]) {
            const int ml = qb * UPB + wave * TM_QPW + lq;
            const float gv = tm_dec(gth_pipe[qb]);
            if (gv > th_lds[ml]) th_lds[ml] = gv;
            gth_pipe[qb] = th_g[guser[qb]];
          }
        }
      }
      __syncthreads();
      // issue the NEXT chunk's global loads AFTER the barrier (a
      // __syncthreads compiles to s_waitcnt vmcnt(0)) so they fly
      // under the MFMA phase
      if (base + CH < it1) load_stg(base + CH);
    }
    if (probe) {
      const unsigned long long now = wall_clock64();
      acc_stage += now - pt;
      pt = now;
    }

    // ---- MFMA: every wave scores the whole chunk for its queries,
    // one 64-query block at a time (QB passes over the staged tile)
#pragma unroll
    for (int qb = 0; qb < QB; ++qb) {
    const int mylist = qb * UPB + wave * TM_QPW + lq;
    float* tvu = topv + mylist * KP;
    int* tiu = topi + mylist * KP;
    f32x4 acc[IFR];
#pragma unroll
    for (int i = 0; i < IFR; ++i) acc[i] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ks = 0; ks < KS; ++ks) {
#pragma unroll
      for (int i = 0; i < IFR; ++i) {
        const int row = i * 16 + lq;           // A operand row = item
        const int col = ks * 64 + lg * 16;     // byte offset of k-slice
        const bf16x8 a = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(yb) +
            row * ROWB + (col ^ ((row & SWM) << 4)));
        acc[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, xf[qb][ks], acc[i], 0, 0, 0);
      }
    }

    // ---- epilogue: common case is one max-reduce + one threshold read
    float mymax = -FLT_MAX;
#pragma unroll
    for (int i = 0; i < IFR; ++i)
#pragma unroll
      for (int r = 0; r < 4; ++r) mymax = fmaxf(mymax, acc[i][r]);
    const float th_b = th_lds[mylist];
    if (has_user[qb] && mymax > th_b) {
      // serialize the query's 4 lane groups: exec-masked blocks of one
      // wave run in program order, and within a group the active lanes
      // all own different queries, so list writes never contend.
      // MUST be a runtime loop with an opaque barrier per iteration:
      // with `#pragma unroll` the four structurally-identical blocks
      // got tail-merged by the compiler into ONE exec-masked block
      // (it does not model cross-lane LDS aliasing), so all groups
      // read the threshold before any insert — losing inserts (the
      // round-2 K=1 test failure: later groups overwrote the max).
#pragma unroll 1
      for (int g = 0; g < 4; ++g) {
        asm volatile("" ::: "memory");  // keep iterations distinct
        if (lg == g) {
          float th = th_lds[mylist];
          // 32-bit tail guard: li < lim replaces the former 64-bit
          // `item < it1` (16 hoisted v_cmp_gt_i64 in the fast loop)
          const int lim = (int)(it1 - base < CH ? it1 - base : CH);
#pragma unroll
          for (int i = 0; i < IFR; ++i) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
              const int li = i * 16 + lg * 4 + r;      // D row
              const long long item = base + li;
              const float s = acc[i][r];
              if (s > th && li < lim) {
                const int gitem = (int)(item + item_base);
                const bool may_ban =
                    ban[qb] != nullptr &&
                    (bloom[qb] >>
                     ((unsigned)((unsigned)gitem * 2654435761u) >> 26)) & 1;
                if ((item_mask == nullptr || !item_mask[item]) &&
                    (!may_ban ||
                     !tm_in_sorted(ban[qb], bn[qb], gitem))) {
                  int mi = 0;
                  float mv = tvu[0];
                  for (int q = 1; q < K; ++q)
                    if (tvu[q] < mv) { mv = tvu[q]; mi = q; }
                  tvu[mi] = s;
                  tiu[mi] = (int)(item + item_base);
                  float nm = tvu[0];
                  for (int q = 1; q < K; ++q) nm = fminf(nm, tvu[q]);
                  th = nm;
                  th_lds[mylist] = nm;
                  // publish to the other slices once the list is full;
                  // test-and-test-and-set: the plain read (L2-hot)
                  // filters publishes already beaten by another slice,
                  // keeping the atomic queue short
                  if (th_g != nullptr && nm > -FLT_MAX &&
                      tm_enc(nm) > th_g[guser[qb]])
                    atomicMax(&th_g[guser[qb]], tm_enc(nm));
                }
              }
            }
          }
        }
      }
    }
    }  // qb
    if (probe) {
      const unsigned long long now = wall_clock64();
      acc_score += now - pt;
      pt = now;
    }
    if (DBUF) {
      // drain chunk ci+1 into the other buffer (its loads were issued
      // one iteration ago), start chunk ci+2's loads, ONE barrier
      if (ci + 1 < n_chunks) {
        drain_to(ybuf(ci + 1));
        if (ci + 2 < n_chunks) load_stg(base + 2 * CH);
      }
      // same pipelined cross-slice threshold fold as the main path
      // (the barrier below orders the th_lds writes)
      if (th_g != nullptr && (ci & 7) == 0 && lg == 0) {
#pragma unroll
        for (int qb = 0; qb < QB; ++qb) {
          if (has_user[qb]) {
            const int ml = qb * UPB + wave * TM_QPW + lq;
            const float gv = tm_dec(gth_pipe[qb]);
            if (gv > th_lds[ml]) th_lds[ml] = gv;
            gth_pipe[qb] = th_g[guser[qb]];
          }
        }
      }
      __syncthreads();
      if (probe) {
        const unsigned long long now = wall_clock64();
        acc_stage += now - pt;
        pt = now;
      }
    }
  }
  __syncthreads();
  if (probe) pt = wall_clock64();

  // ---- write out: one candidate group per slice per query
  for (int e = tid; e < UPBT * K; e += BS) {
    const int list = e / K;
    const int q = e % K;
    const long long gu = u0 + list;
    if (gu < B) {
      const long long o = (gu * n_slices + slice) * (long long)K + q;
      out_val[o] = topv[list * KP + q];
      out_idx[o] = topi[list * KP + q];
    }
  }
  if (probe) {
    atomicAdd(&prof[0], acc_setup);
    atomicAdd(&prof[1], acc_stage);
    atomicAdd(&prof[2], acc_score);
    atomicAdd(&prof[3], wall_clock64() - pt);
    atomicAdd(&prof[4], 1ull);
  }
}

// 8-wave (128 queries/WG) wide variant: halves the Y stream per query
// (each staged slice serves 2x the queries). Only meaningful for
// F >= 64 (the staging granule math needs CH*F*2/16 >= block size).
template <int FF>
static void launch_topk_mfma_wide(
    dim3 grid, size_t lds_bytes, hipStream_t stream,
    const unsigned short* Xq, const unsigned short* Y,
    const uint8_t* item_mask, const long long* ban_indptr,
    const int* ban_indices, float* out_val, int* out_idx,
    int B, long long N, int K, int n_slices, int item_base,
    unsigned* th_g, int kflags) {
  if constexpr (FF >= 64) {
    hipLaunchKernelGGL(
        (topk_mfma_kernel<FF, false, false, TM_CHUNK, 8>), grid,
        dim3(512), lds_bytes, stream, Xq, Y, item_mask, ban_indptr,
        ban_indices, out_val, out_idx, B, N, K, n_slices, item_base,
        nullptr, th_g, kflags);
  }
}

extern "C" void launch_topk_mfma(
    const unsigned short* Xq, const unsigned short* Y,
    const uint8_t* item_mask, const long long* ban_indptr,
    const int* ban_indices, float* out_val, int* out_idx,
    int B, long long N, int f, int K, int n_slices, int item_base,
    unsigned long long* prof, unsigned* th_g, hipStream_t stream)
{
  dim3 grid((B + TM_UPB - 1) / TM_UPB, n_slices);
  dim3 block(256);
  dim3 grid_w((B + 127) / 128, n_slices);
  const char* e_db = getenv("PIO_TOPK_DB");
  const bool use_db = e_db != nullptr && e_db[0] == '1';
  const char* e_ch = getenv("PIO_TOPK_CHUNK");
  const int chunk = (e_ch && atoi(e_ch) == 128) ? 128 : TM_CHUNK;
  const char* e_g = getenv("PIO_TOPK_GLL");
  const bool use_gll = e_g != nullptr && e_g[0] == '1';
  const char* e_w = getenv("PIO_TOPK_WIDE");
  const bool use_wide = e_w != nullptr && e_w[0] == '1' && f >= 64 &&
                        prof == nullptr && !use_db && chunk == TM_CHUNK;
  const char* e_bl = getenv("PIO_TOPK_BLOOM");
  const int kflags = (e_bl != nullptr && e_bl[0] == '0') ? 1 : 0;
  const char* e_q = getenv("PIO_TOPK_QB");
  const bool use_qb2 = e_q != nullptr && e_q[0] == '2' && !use_db &&
                       !use_gll && !use_wide && chunk == TM_CHUNK;
  dim3 grid_q2((B + 2 * TM_UPB - 1) / (2 * TM_UPB), n_slices);
#define LAUNCH_M(FF)                                                         \
  do {                                                                       \
    size_t lds_bytes = (size_t)(use_db ? 2 : 1) * chunk * (FF * 2) +         \
                       (sizeof(float) + sizeof(int)) *                       \
                           (use_qb2 ? 2 : 1) * TM_UPB * (K + 1) +            \
                       sizeof(float) * (use_qb2 ? 2 : 1) * TM_UPB;           \
    static bool attr_set_##FF = false;                                       \
    if (!attr_set_##FF && lds_bytes > 64 * 1024) {                           \
      hipFuncSetAttribute(                                                   \
          reinterpret_cast<const void*>(&topk_mfma_kernel<FF, false, false>),       \
          hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);           \
      hipFuncSetAttribute(                                                   \
          reinterpret_cast<const void*>(&topk_mfma_kernel<FF, true, false>),        \
          hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);           \
      hipFuncSetAttribute(                                                   \
          reinterpret_cast<const void*>(                                     \
              &topk_mfma_kernel<FF, false, false, TM_CHUNK, TM_WAVES,        \
                                false, 2>),                                  \
          hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);           \
      hipFuncSetAttribute(                                                   \
          reinterpret_cast<const void*>(                                     \
              &topk_mfma_kernel<FF, true, false, TM_CHUNK, TM_WAVES,         \
                                false, 2>),                                  \
          hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);           \
      attr_set_##FF = true;                                                  \
    }                                                                        \
    if (use_qb2) {                                                           \
      if (prof != nullptr)                                                   \
        hipLaunchKernelGGL(                                                  \
            (topk_mfma_kernel<FF, true, false, TM_CHUNK, TM_WAVES, false,    \
                              2>),                                           \
            grid_q2, block, lds_bytes, stream, Xq, Y, item_mask, ban_indptr, \
            ban_indices, out_val, out_idx, B, N, K, n_slices, item_base,     \
            prof, th_g, kflags);                                                           \
      else                                                                   \
        hipLaunchKernelGGL(                                                  \
            (topk_mfma_kernel<FF, false, false, TM_CHUNK, TM_WAVES, false,   \
                              2>),                                           \
            grid_q2, block, lds_bytes, stream, Xq, Y, item_mask, ban_indptr, \
            ban_indices, out_val, out_idx, B, N, K, n_slices, item_base,     \
            nullptr, th_g, kflags);                                                        \
    } else if (prof != nullptr) {                                            \
      if (use_db)                                                            \
        hipLaunchKernelGGL((topk_mfma_kernel<FF, true, true>), grid, block,  \
                           lds_bytes, stream, Xq, Y, item_mask, ban_indptr,  \
                           ban_indices, out_val, out_idx, B, N, K, n_slices, \
                           item_base, prof, th_g, kflags);                                 \
      else if (chunk == 128)                                                 \
        hipLaunchKernelGGL((topk_mfma_kernel<FF, true, false, 128>), grid,   \
                           block, lds_bytes, stream, Xq, Y, item_mask,       \
                           ban_indptr, ban_indices, out_val, out_idx, B, N,  \
                           K, n_slices, item_base, prof, th_g, kflags);                    \
      else                                                                   \
        hipLaunchKernelGGL((topk_mfma_kernel<FF, true, false>), grid, block, \
                           lds_bytes, stream, Xq, Y, item_mask, ban_indptr,  \
                           ban_indices, out_val, out_idx, B, N, K, n_slices, \
                           item_base, prof, th_g, kflags);                                 \
    } else if (use_wide) {                                                   \
      size_t lds_w = (size_t)TM_CHUNK * (FF * 2) +                           \
                     (sizeof(float) + sizeof(int)) * 128 * (K + 1) +         \
                     sizeof(float) * 128;                                    \
      launch_topk_mfma_wide<FF>(grid_w, lds_w, stream, Xq, Y, item_mask,     \
                                ban_indptr, ban_indices, out_val, out_idx,   \
                                B, N, K, n_slices, item_base, th_g,          \
                                kflags);                                     \
    } else {                                                                 \
      if (use_db)                                                            \
        hipLaunchKernelGGL((topk_mfma_kernel<FF, false, true>), grid, block, \
                           lds_bytes, stream, Xq, Y, item_mask, ban_indptr,  \
                           ban_indices, out_val, out_idx, B, N, K, n_slices, \
                           item_base, nullptr, th_g, kflags);                              \
      else if (chunk == 128)                                                 \
        hipLaunchKernelGGL((topk_mfma_kernel<FF, false, false, 128>), grid,  \
                           block, lds_bytes, stream, Xq, Y, item_mask,       \
                           ban_indptr, ban_indices, out_val, out_idx, B, N,  \
                           K, n_slices, item_base, nullptr, th_g, kflags);                 \
      else if (use_gll)                                                      \
        hipLaunchKernelGGL(                                                  \
            (topk_mfma_kernel<FF, false, false, TM_CHUNK, TM_WAVES, true>),  \
            grid, block, lds_bytes, stream, Xq, Y, item_mask, ban_indptr,    \
            ban_indices, out_val, out_idx, B, N, K, n_slices, item_base,     \
            nullptr, th_g, kflags);                                                        \
      else                                                                   \
        hipLaunchKernelGGL((topk_mfma_kernel<FF, false, false>), grid,       \
                           block, lds_bytes, stream, Xq, Y, item_mask,       \
                           ban_indptr, ban_indices, out_val, out_idx, B, N,  \
                           K, n_slices, item_base, nullptr, th_g, kflags);                 \
    }                                                                        \
  } while (0)
  switch (f) {
    case 32: LAUNCH_M(32); break;
    case 64: LAUNCH_M(64); break;
    case 128: LAUNCH_M(128); break;
    default: break;
  }
#undef LAUNCH_M
}
