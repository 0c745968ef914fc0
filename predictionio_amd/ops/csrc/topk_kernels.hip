// Fused masked top-K scoring for MI355X (gfx950, CDNA4).
//
// Replaces the reference's serve-time scoring loops:
//  - recommendation `recommendProductsWithFilter`: blas.ddot per item +
//    scored.top(num) (examples/.../ALSModel.scala:44-60)
//  - similarproduct cosine + PriorityQueue top-N
//    (examples/.../ALSAlgorithm.scala:168-242)
//  - ecommerce predictKnownUser / predictSimilar
//    (examples/.../ECommAlgorithm.scala:471-506, 541-599)
//
// Design (v3, MI355X-first): scores = Xq . Y^T fused with masking and an
// on-device top-K reduction, never materializing the B x N score matrix
// (N up to 10^7). Grid = (item_slices, user_blocks of 64).
//
//  - LANE = USER: each lane carries its user's query vector in packed
//    float2 registers (loaded once per workgroup) and a scalar running
//    K-th-best threshold. Insertions go to the lane's own list — fully
//    parallel, no cross-lane serialization. (v1 had 8 users/WG and
//    re-read Y B/8 times from HBM — 1.3 TB per 4096-query batch; v2
//    tiled 64 users but kept thresholds in a register array indexed by a
//    16-iteration user loop, forcing a 16x unroll that blew L1I.)
//  - 64-item chunks of Y staged through LDS (coalesced); each wave scores
//    its own 16-item quarter: per item the k-loop is an LDS-broadcast
//    float2 read against the xreg registers — v_pk_fma_f32 packed math,
//    two independent accumulator chains.
//  - per-(wave, user) top-K lists in LDS; the 4 waves are independent
//    candidate groups, so the output carries n_slices*4 groups of K and
//    the host-side merge (one small torch.topk) folds them.
//
// Masks:
//  - item_mask: optional uint8[N], 1 = globally banned (uniform per item,
//    checked once per wave)
//  - per-user banned list (seen/blacklisted items): CSR int32; the owning
//    lane binary-searches its own list per candidate insertion.

#include <float.h>
#include <hip/hip_runtime.h>

#define TK_CHUNK 32   // items staged per LDS pass (32: fits 3 WGs/CU with K=20 lists)
#define TK_MAXK 64    // max supported K
#define TK_WAVES 4
#define TK_UPB 64     // users per block (= lane count)

typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ bool in_sorted(const int* arr, int n, int x) {
  int lo = 0, hi = n - 1;
  while (lo <= hi) {
    int mid = (lo + hi) >> 1;
    int v = arr[mid];
    if (v == x) return true;
    if (v < x) lo = mid + 1; else hi = mid - 1;
  }
  return false;
}

// PROF=true compiles in wall_clock64() phase accumulators (setup / item
// staging / score+insert / writeback), summed per workgroup into
// prof[5] = {setup, stage, score, write, blocks}. The live serving path
// launches the PROF=false instantiation — zero probe cost.
template <int F, bool PROF>
__global__ __launch_bounds__(256, 3) void topk_score_kernel(
    const float* __restrict__ Xq,        // B x F query vectors
    const float* __restrict__ Y,         // N x F item factors
    const uint8_t* __restrict__ item_mask,       // N or nullptr
    const long long* __restrict__ ban_indptr,    // B+1 or nullptr
    const int* __restrict__ ban_indices,         // sorted per user
    float* __restrict__ out_val,         // B x (n_slices*TK_WAVES) x K
    int* __restrict__ out_idx,
    int B, long long N, int K, int n_slices, int item_base,
    unsigned long long* prof)
{
  constexpr int FP = F + 4;  // row stride: 16B-aligned so the score loop
                             // reads ys rows as ds_read_b128 quads
  // dynamic LDS: ys[TK_CHUNK][FP] | topv[TK_WAVES*TK_UPB][K] | topi[...]
  extern __shared__ float lds[];
  float* ys = lds;
  float* topv = ys + TK_CHUNK * FP;
  int* topi = reinterpret_cast<int*>(topv + TK_WAVES * TK_UPB * (K + 1));

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  // K-list row stride padded by one: lane-consecutive K=20 lists had an
  // 8-way LDS bank conflict (gcd(20,32)=4; PMC: 485k conflicts/wave =
  // 34% of LDS instructions); stride 21 is coprime with the 32 banks.
  const int KP = K + 1;

  const bool probe = PROF && tid == 0;
  unsigned long long pt = 0, acc_setup = 0, acc_stage = 0, acc_score = 0;
  if (probe) pt = wall_clock64();

  const int slice = blockIdx.x;
  const int ublock = blockIdx.y;
  const long long u0 = (long long)ublock * TK_UPB;
  const long long guser = u0 + lane;
  const bool has_user = guser < B;

  // item range of this slice
  const long long per = (N + n_slices - 1) / n_slices;
  const long long it0 = (long long)slice * per;
  const long long it1 = min(N, it0 + per);

  // ---- load query vectors: coalesced stage into ys, transpose to regs.
  // NOTE: TK_UPB (64) rows are staged but ys is TK_CHUNK (32) rows — the
  // upper rows intentionally overflow into the top-list region, which is
  // initialized only AFTER the transpose below, and the total dynamic
  // allocation covers it (x staging 16.9 KB < 49.4 KB).
  for (int e = tid; e < TK_UPB * F; e += 256) {
    const int u = e / F;
    const int k = e % F;
    ys[u * FP + k] = (u0 + u < B) ? Xq[(u0 + u) * F + k] : 0.f;
  }
  __syncthreads();
  f32x2 xreg[F / 2];
  {
    const float* xrow = ys + lane * FP;
#pragma unroll
    for (int q = 0; q < F / 2; ++q)
      xreg[q] = f32x2{xrow[2 * q], xrow[2 * q + 1]};
  }
  __syncthreads();

  // ---- init this block's top-K lists
  for (int e = tid; e < TK_WAVES * TK_UPB * KP; e += 256) {
    topv[e] = -FLT_MAX;
    topi[e] = -1;
  }
  __syncthreads();

  if (probe) {
    const unsigned long long now = wall_clock64();
    acc_setup = now - pt;
    pt = now;
  }

  float th = -FLT_MAX;               // this lane's K-th best (this wave)
  float* tvu = topv + (wave * TK_UPB + lane) * KP;
  int* tiu = topi + (wave * TK_UPB + lane) * KP;

  // this lane's banned list
  const int* ban = nullptr;
  int bn = 0;
  if (ban_indptr != nullptr && has_user) {
    const long long b0 = ban_indptr[guser];
    bn = (int)(ban_indptr[guser + 1] - b0);
    ban = ban_indices + b0;
  }

  // Item staging is software-pipelined through registers: while chunk i
  // is scored out of LDS, chunk i+1's global loads are already in flight
  // into stg[] (the phase probe measured un-overlapped staging at 33.6%
  // of WG time — profiles/serve_phase_probe_r1.txt). Costs SREG = F/8
  // VGPRs and no extra LDS. F=128's xreg pressure (128 VGPRs) cannot
  // afford the pipeline registers, so it keeps the direct stage.
  // float4 loads (Y rows are 16B-aligned) keep the in-flight register
  // count low: a scalar-element pipeline spilled 68 B at F=64 — the 8
  // loads' address registers outlived the fully unrolled score loop.
  constexpr int Q4 = (TK_CHUNK * F) / 4;              // float4s per chunk
  constexpr int SREG4 = Q4 >= 256 ? Q4 / 256 : 1;
  constexpr bool PIPE = (F <= 64);
  f32x4 stg4[PIPE ? SREG4 : 1];
  if (PIPE) {
    const int cn0 = (int)min((long long)TK_CHUNK, it1 - it0);
    // uniform chunk base folded into the pointer: the per-thread offset
    // stays 32-bit (c*F+k < 8K), so the loads take the scalar-base +
    // v-offset form instead of spilling a 64-bit vector address
    const float* Yb = Y + it0 * (long long)F;
#pragma unroll
    for (int r = 0; r < (PIPE ? SREG4 : 1); ++r) {
      const int e4 = tid + r * 256;
      const int c = e4 / (F / 4);
      const int k = (e4 % (F / 4)) * 4;
      stg4[r] = (e4 < Q4 && c < cn0)
          ? *reinterpret_cast<const f32x4*>(&Yb[c * F + k])
          : f32x4{0.f, 0.f, 0.f, 0.f};
    }
  }

  for (long long base = it0; base < it1; base += TK_CHUNK) {
    const int cn = (int)min((long long)TK_CHUNK, it1 - base);
    __syncthreads();
    if (PIPE) {
      // drain the in-flight registers into LDS (component stores: the
      // FP=F+2 row stride is 8B- but not 16B-aligned), then issue the
      // next chunk's loads so they fly during the score pass
#pragma unroll
      for (int r = 0; r < (PIPE ? SREG4 : 1); ++r) {
        const int e4 = tid + r * 256;
        if (e4 < Q4) {
          const int c = e4 / (F / 4);
          const int k = (e4 % (F / 4)) * 4;
          float* dst = ys + c * FP + k;
          dst[0] = stg4[r].x;
          dst[1] = stg4[r].y;
          dst[2] = stg4[r].z;
          dst[3] = stg4[r].w;
        }
      }
    } else {
      // coalesced stage of cn item rows
      for (int e = tid; e < cn * F; e += 256) {
        const int c = e / F;
        const int k = e % F;
        ys[c * FP + k] = Y[(base + c) * F + k];
      }
    }
    __syncthreads();
    if (PIPE) {
      // issue the NEXT chunk's loads AFTER the barrier: __syncthreads
      // compiles to s_waitcnt vmcnt(0), so loads issued before it were
      // drained at the barrier (the phase probe's "stage = inter-wave
      // barrier wait" signature). Here they fly under the score loop
      // and are awaited by the next iteration's first barrier.
      const long long nbase = base + TK_CHUNK;
      if (nbase < it1) {
        const int cnn = (int)min((long long)TK_CHUNK, it1 - nbase);
        const float* Yb = Y + nbase * (long long)F;
#pragma unroll
        for (int r = 0; r < (PIPE ? SREG4 : 1); ++r) {
          const int e4 = tid + r * 256;
          const int c = e4 / (F / 4);
          const int k = (e4 % (F / 4)) * 4;
          stg4[r] = (e4 < Q4 && c < cnn)
              ? *reinterpret_cast<const f32x4*>(&Yb[c * F + k])
              : f32x4{0.f, 0.f, 0.f, 0.f};
        }
      }
    }
    if (probe) {
      const unsigned long long now = wall_clock64();
      acc_stage += now - pt;
      pt = now;
    }

    // wave w scores items [8w, 8w+8) of the chunk for all 64 users, in
    // GROUPS OF 4: serve PMC showed ~370 wait cycles per item — the
    // per-item threshold branch serialized the next item's LDS reads
    // behind the current item's score. Four interleaved dot chains keep
    // 4 items' reads in flight; the (rare) mask/ban checks move into the
    // insert path.
    const int c_lo = wave * (TK_CHUNK / TK_WAVES);
    const int c_hi = min(cn, c_lo + TK_CHUNK / TK_WAVES);
    // GRP=2 fits the 12-waves/CU register budget at F<=64; F=128's
    // xreg alone is 128 VGPRs, so it stays at one chain.
    constexpr int GRP = F <= 64 ? 2 : 1;
    for (int c0 = c_lo; c0 < c_hi; c0 += GRP) {
      const int gn = min(GRP, c_hi - c0);
      f32x2 a0 = {0.f, 0.f}, a1 = {0.f, 0.f};
      const f32x4* yr0 = reinterpret_cast<const f32x4*>(ys + (c0 + 0) * FP);
      const f32x4* yr1 = reinterpret_cast<const f32x4*>(
          ys + (c0 + (GRP > 1 ? 1 : 0)) * FP);
      // FULLY unrolled: q must be a literal so xreg[] stays in registers
      // (a capped unroll sent it to scratch). ys rows are read in b128
      // quads: halves the LDS instruction count in a loop that is
      // ISSUE-bound (LDS reads and v_pk_fma share the SIMD issue slot).
#pragma unroll
      for (int q = 0; q < F / 4; ++q) {
        const f32x4 y0 = yr0[q];
        a0 += xreg[2 * q] * f32x2{y0.x, y0.y};
        a0 += xreg[2 * q + 1] * f32x2{y0.z, y0.w};
        if (GRP > 1) {
          const f32x4 y1 = yr1[q];
          a1 += xreg[2 * q] * f32x2{y1.x, y1.y};
          a1 += xreg[2 * q + 1] * f32x2{y1.z, y1.w};
        }
      }
      float sg[2] = {a0.x + a0.y, a1.x + a1.y};
#pragma unroll
      for (int g = 0; g < GRP; ++g) {
      const int c = c0 + g;
      const long long item = base + c;
      const float s = g < gn ? sg[g] : -FLT_MAX;
      if (has_user && s > th) {
        if ((item_mask == nullptr || !item_mask[item]) &&
            (ban == nullptr ||
             !in_sorted(ban, bn, (int)(item + item_base)))) {
          // replace current min of this lane's K-list
          int mi = 0;
          float mv = tvu[0];
          for (int q = 1; q < K; ++q)
            if (tvu[q] < mv) { mv = tvu[q]; mi = q; }
          tvu[mi] = s;
          tiu[mi] = (int)(item + item_base);
          float nm = tvu[0];
          for (int q = 1; q < K; ++q) nm = fminf(nm, tvu[q]);
          th = nm;
        }
      }
      }
    }
    if (probe) {
      const unsigned long long now = wall_clock64();
      acc_score += now - pt;
      pt = now;
    }
  }
  __syncthreads();
  if (probe) pt = wall_clock64();

  // ---- write out: group g = slice*TK_WAVES + wave, K entries per user
  for (int e = tid; e < TK_WAVES * TK_UPB * K; e += 256) {
    const int w = e / (TK_UPB * K);
    const int u = (e / K) % TK_UPB;
    const int q = e % K;
    const long long gu = u0 + u;
    if (gu < B) {
      const long long g = (long long)slice * TK_WAVES + w;
      const long long o = (gu * n_slices * TK_WAVES + g) * K + q;
      out_val[o] = topv[(w * TK_UPB + u) * KP + q];
      out_idx[o] = topi[(w * TK_UPB + u) * KP + q];
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

extern "C" void launch_topk_score(
    const float* Xq, const float* Y, const uint8_t* item_mask,
    const long long* ban_indptr, const int* ban_indices,
    float* out_val, int* out_idx,
    int B, long long N, int f, int K, int n_slices, int item_base,
    unsigned long long* prof, hipStream_t stream)
{
  dim3 grid(n_slices, (B + TK_UPB - 1) / TK_UPB);
  dim3 block(256);
#define LAUNCH(FF)                                                          \
  do {                                                                      \
    size_t lds_bytes =                                                      \
        sizeof(float) * (TK_CHUNK * (FF + 4)) +                             \
        (sizeof(float) + sizeof(int)) * TK_WAVES * TK_UPB * (K+1);        \
    /* small-K floor: the query-staging pass writes TK_UPB rows     */  \
    const size_t stage_need = sizeof(float) * TK_UPB * (FF + 4);        \
    if (lds_bytes < stage_need) lds_bytes = stage_need;                 \
    static bool attr_set_##FF = false;                                      \
    if (!attr_set_##FF && lds_bytes > 64 * 1024) {                          \
      hipFuncSetAttribute(                                                  \
          reinterpret_cast<const void*>(&topk_score_kernel<FF, false>),     \
          hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);          \
      hipFuncSetAttribute(                                                  \
          reinterpret_cast<const void*>(&topk_score_kernel<FF, true>),      \
          hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);          \
      attr_set_##FF = true;                                                 \
    }                                                                       \
    if (prof != nullptr)                                                    \
      hipLaunchKernelGGL((topk_score_kernel<FF, true>), grid, block,        \
                         lds_bytes, stream, Xq, Y, item_mask, ban_indptr,   \
                         ban_indices, out_val, out_idx, B, N, K, n_slices,  \
                         item_base, prof);                                  \
    else                                                                    \
      hipLaunchKernelGGL((topk_score_kernel<FF, false>), grid, block,       \
                         lds_bytes, stream, Xq, Y, item_mask, ban_indptr,   \
                         ban_indices, out_val, out_idx, B, N, K, n_slices,  \
                         item_base, nullptr);                               \
  } while (0)
  switch (f) {
    case 16: LAUNCH(16); break;
    case 32: LAUNCH(32); break;
    case 64: LAUNCH(64); break;
    case 128: LAUNCH(128); break;
    default: break;
  }
#undef LAUNCH
}
