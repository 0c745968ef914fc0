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
// Design (v4, MI355X-first): scores = Xq . Y^T fused with masking and an
// on-device top-K reduction, never materializing the full B x N score
// matrix (N up to 10^7). Grid = (item_slices, user_blocks of 64).
//
// Per 64-item chunk each workgroup runs two phases:
//  PHASE A (GEMM tile): the 64-user x 64-item score tile is computed with
//    register tiling — each lane owns a 4-user x 4-item accumulator tile
//    and reads x/y in float4 k-quads, so LDS traffic is 2 B/MAC and the
//    loop is VALU-bound (v_pk_fma_f32). Earlier versions were
//    LDS-broadcast-bound at ~1 read/MAC: v1 re-read Y B/8 times from HBM
//    (1.3 TB/batch, 656 ms); v2's 16x-unrolled user loop blew L1I
//    (1129 ms); v3 (lane-per-user broadcast dot) measured 293 ms — the
//    analytic VALU bound is ~32 ms.
//  PHASE B (reduce): scores land in an LDS tile; wave w owns users
//    [16w, 16w+16) — one lane per user scans the full 64-item row and
//    inserts into the user's single top-K list. No cross-lane or
//    cross-wave sharing, so lists are per-user (TK_UPB*K LDS) and the
//    output carries n_slices groups of K.
//
// Masks:
//  - item_mask: optional uint8[N], 1 = globally banned (checked once per
//    item in phase B)
//  - per-user banned list (seen/blacklisted): CSR int32; the owning lane
//    binary-searches its own sorted list per candidate insertion.

#include <float.h>
#include <hip/hip_runtime.h>

#define TK_CHUNK 64   // items per tile
#define TK_MAXK 64    // max supported K
#define TK_WAVES 4
#define TK_UPB 64     // users per block

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

template <int F>
__global__ __launch_bounds__(256) void topk_score_kernel(
    const float* __restrict__ Xq,        // B x F query vectors
    const float* __restrict__ Y,         // N x F item factors
    const uint8_t* __restrict__ item_mask,       // N or nullptr
    const long long* __restrict__ ban_indptr,    // B+1 or nullptr
    const int* __restrict__ ban_indices,         // sorted per user
    float* __restrict__ out_val,         // B x (n_slices*TK_WAVES) x K
    int* __restrict__ out_idx,
    int B, long long N, int K, int n_slices, int item_base)
{
  constexpr int FP = F + 4;   // factor row stride (float4-aligned, padded)
  constexpr int SP = TK_CHUNK + 4;  // score row stride
  // dynamic LDS: xs[TK_UPB][FP] | ys[TK_CHUNK][FP] | sc[TK_UPB][SP] |
  //              topv[TK_UPB][K] | topi[...]
  extern __shared__ float lds[];
  float* xs = lds;
  float* ys = xs + TK_UPB * FP;
  float* sc = ys + TK_CHUNK * FP;
  float* topv = sc + TK_UPB * SP;
  int* topi = reinterpret_cast<int*>(topv + TK_UPB * K);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  const int slice = blockIdx.x;
  const int ublock = blockIdx.y;
  const long long u0 = (long long)ublock * TK_UPB;

  // phase-A tile coords: lane = ug*16 + ig → users [4*ug..+4) of this
  // wave's 16-user group, items {ig, ig+16, ig+32, ig+48}. Items stride
  // 16 (not 4): with row stride 68 floats, rows 4 apart land on only two
  // bank positions (68*4*4 B ≡ 16 mod 32 banks → 8-way conflict on the
  // y b128 reads, measured 361 ms vs v3's 293); rows 16 apart spread
  // over 8 positions (2-way).
  const int ug = lane >> 4;
  const int ig = lane & 15;
  const int urow = wave * 16 + ug * 4;   // first of this lane's 4 users
  const int icol = ig;                   // lane's items: icol + 16*i

  const long long per = (N + n_slices - 1) / n_slices;
  const long long it0 = (long long)slice * per;
  const long long it1 = min(N, it0 + per);

  // ---- stage the 64 query vectors once ----
  for (int e = tid; e < TK_UPB * F; e += 256) {
    const int u = e / F;
    const int k = e % F;
    xs[u * FP + k] = (u0 + u < B) ? Xq[(u0 + u) * F + k] : 0.f;
  }
  for (int e = tid; e < TK_UPB * K; e += 256) {
    topv[e] = -FLT_MAX;
    topi[e] = -1;
  }
  __syncthreads();

  // phase-B ownership: wave w, lane < 16 → user 16w + lane
  const int u_own = wave * 16 + lane;
  const bool owner = lane < 16;
  float th = -FLT_MAX;               // owned user's running K-th best
  float* tvu = topv + u_own * K;
  int* tiu = topi + u_own * K;

  const int* ban = nullptr;
  int bn = 0;
  const long long guser = u0 + u_own;
  const bool has_user = owner && guser < B;
  if (ban_indptr != nullptr && has_user) {
    const long long b0 = ban_indptr[guser];
    bn = (int)(ban_indptr[guser + 1] - b0);
    ban = ban_indices + b0;
  }

  for (long long base = it0; base < it1; base += TK_CHUNK) {
    const int cn = (int)min((long long)TK_CHUNK, it1 - base);
    __syncthreads();
    // coalesced stage of cn item rows (pad rows zero so phase A is
    // branch-free; their scores are discarded in phase B)
    for (int e = tid; e < TK_CHUNK * F; e += 256) {
      const int c = e / F;
      const int k = e % F;
      ys[c * FP + k] = c < cn ? Y[(base + c) * F + k] : 0.f;
    }
    __syncthreads();

    // ---- PHASE A: 4x4 register tile over k in float4 quads ----
    {
      f32x4 acc[4];  // acc[i] = scores of item icol+i for the 4 users
#pragma unroll
      for (int i = 0; i < 4; ++i) acc[i] = f32x4{0.f, 0.f, 0.f, 0.f};
      const float* xrow0 = xs + (urow + 0) * FP;
      const float* xrow1 = xs + (urow + 1) * FP;
      const float* xrow2 = xs + (urow + 2) * FP;
      const float* xrow3 = xs + (urow + 3) * FP;
      const float* yrow0 = ys + (icol + 0) * FP;
      const float* yrow1 = ys + (icol + 16) * FP;
      const float* yrow2 = ys + (icol + 32) * FP;
      const float* yrow3 = ys + (icol + 48) * FP;
      // unroll capped: full unroll kept 16 k-quads of x/y live and pushed
      // the kernel to 334 VGPRs (1 wave/SIMD)
#pragma unroll 2
      for (int k = 0; k < F; k += 4) {
        const f32x4 x0 = *reinterpret_cast<const f32x4*>(xrow0 + k);
        const f32x4 x1 = *reinterpret_cast<const f32x4*>(xrow1 + k);
        const f32x4 x2 = *reinterpret_cast<const f32x4*>(xrow2 + k);
        const f32x4 x3 = *reinterpret_cast<const f32x4*>(xrow3 + k);
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const float* yr = i == 0 ? yrow0 : i == 1 ? yrow1
                            : i == 2 ? yrow2 : yrow3;
          const f32x4 y4 = *reinterpret_cast<const f32x4*>(yr + k);
          // acc[i].j += x_j[k..k+4] . y[k..k+4] — packed fp32 fma
          acc[i].x += x0.x * y4.x + x0.y * y4.y + x0.z * y4.z + x0.w * y4.w;
          acc[i].y += x1.x * y4.x + x1.y * y4.y + x1.z * y4.z + x1.w * y4.w;
          acc[i].z += x2.x * y4.x + x2.y * y4.y + x2.z * y4.z + x2.w * y4.w;
          acc[i].w += x3.x * y4.x + x3.y * y4.y + x3.z * y4.z + x3.w * y4.w;
        }
      }
      // write the 4x4 tile
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        sc[(urow + 0) * SP + icol + 16 * i] = acc[i].x;
        sc[(urow + 1) * SP + icol + 16 * i] = acc[i].y;
        sc[(urow + 2) * SP + icol + 16 * i] = acc[i].z;
        sc[(urow + 3) * SP + icol + 16 * i] = acc[i].w;
      }
    }
    __syncthreads();

    // ---- PHASE B: one lane per owned user scans the full chunk ----
    if (has_user) {
      const int c_lo = 0;
      const int c_hi = cn;
      const float* srow = sc + u_own * SP;
      for (int c = c_lo; c < c_hi; ++c) {
        const float s = srow[c];
        if (s <= th) continue;
        const long long item = base + c;
        if (item_mask != nullptr && item_mask[item]) continue;
        if (ban != nullptr &&
            in_sorted(ban, bn, (int)(item + item_base))) continue;
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
  __syncthreads();

  // ---- write out: K entries per user for this slice
  for (int e = tid; e < TK_UPB * K; e += 256) {
    const int u = e / K;
    const int q = e % K;
    const long long gu = u0 + u;
    if (gu < B) {
      const long long o = (gu * n_slices + (long long)slice) * K + q;
      out_val[o] = topv[u * K + q];
      out_idx[o] = topi[u * K + q];
    }
  }
}

extern "C" void launch_topk_score(
    const float* Xq, const float* Y, const uint8_t* item_mask,
    const long long* ban_indptr, const int* ban_indices,
    float* out_val, int* out_idx,
    int B, long long N, int f, int K, int n_slices, int item_base,
    hipStream_t stream)
{
  dim3 grid(n_slices, (B + TK_UPB - 1) / TK_UPB);
  dim3 block(256);
#define LAUNCH(FF)                                                          \
  do {                                                                      \
    const size_t lds_bytes =                                                \
        sizeof(float) * ((TK_UPB + TK_CHUNK) * (FF + 4) +                   \
                         TK_UPB * (TK_CHUNK + 4)) +                         \
        (sizeof(float) + sizeof(int)) * TK_UPB * K;                         \
    static bool attr_set_##FF = false;                                      \
    if (!attr_set_##FF && lds_bytes > 64 * 1024) {                          \
      hipFuncSetAttribute(                                                  \
          reinterpret_cast<const void*>(&topk_score_kernel<FF>),            \
          hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);          \
      attr_set_##FF = true;                                                 \
    }                                                                       \
    hipLaunchKernelGGL((topk_score_kernel<FF>), grid, block, lds_bytes,     \
                       stream, Xq, Y, item_mask, ban_indptr, ban_indices,   \
                       out_val, out_idx, B, N, K, n_slices, item_base);     \
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
