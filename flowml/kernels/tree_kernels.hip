// CDNA4 (gfx950) kernels for the tree-ensemble hot path.
//
// K4a  bin_codes        — quantile binning: f32 features -> uint8 bin codes
// K2   tree_hist_cls    — per-node class histograms, ALL trees in one pass:
//                         one workgroup owns (tree × row-tile), privatizes
//                         that tree's frontier histograms in LDS (float
//                         atomics; counts are integers so sums are exact and
//                         run-to-run deterministic), flushes with global
//                         atomics. Optionally fuses the K4b level advance
//                         (nodeid updated in place) and computes BaggedPoint
//                         weights from a splitmix64 hash (no bag tensor).
//                         The #1 hot loop of the flagship config.
// K2b  tree_hist_reg    — regression variant: 2-wide (Σwt, Σw) cells + one
//                         per-node Σwt² slot (variance gain needs no per-bin
//                         Σt²) for GBT trees
// K4b  node_advance     — standalone advance (CPU-parity and GBT paths)
// K8   tree_predict     — batched ensemble inference: sum of leaf vectors
// K1   logreg_delta     — fused logits+softmax+delta+loss for LogisticRegression
// K3   split_scan        — classification split-gain scan (packed rows)
// K3b  split_scan_reg    — variance scan on the 2-wide regression layout
// K6   softmax_xent     — row-wise softmax/cross-entropy loss + delta
//
// Node identifiers are GLOBAL arena indices (one struct-of-arrays node store
// for the whole forest — flowml/classification/tree.py), so the node->group
// lut is 1-D and the advance arrays need no per-tree offsets.
//
// Grid layout: blockIdx.x = tree, blockIdx.y = row tile — consecutive blocks
// process the SAME codes tile for different trees, so the 672 KB tile stays
// in the XCD's 4 MB L2 across all 100 trees instead of streaming the whole
// codes matrix from HBM once per tree.
//
// Design notes (per /opt/skills guides): wave64; blocks are multiples of 64;
// 512-thread blocks so 8 waves share one LDS histogram copy (16 waves/CU at
// the 60 KB deep-level histograms); LDS histogram capped at 128 KiB by the
// host-side chunker (_hist_lds_cap).

#include <hip/hip_runtime.h>
#include <limits.h>
#include <stdint.h>
#include <stdlib.h>

// launch-shape knobs (read once; for on-box tuning sweeps without rebuilds)
static int flowml_env_int(const char* name, int dflt) {
  const char* v = getenv(name);
  return v ? atoi(v) : dflt;
}

#define FLOWML_MAXK 16  // max classes handled in-register by rowwise kernels

// ---------------------------------------------------------------------------
// K4a — binning
// ---------------------------------------------------------------------------

extern "C" __global__ void flowml_bin_codes(
    const float* __restrict__ X, const float* __restrict__ thr,
    uint8_t* __restrict__ out, long long N, int F, int TB) {
  extern __shared__ float s_thr[];
  for (int i = threadIdx.x; i < F * TB; i += blockDim.x) s_thr[i] = thr[i];
  __syncthreads();
  const long long total = N * (long long)F;
  for (long long idx = blockIdx.x * (long long)blockDim.x + threadIdx.x;
       idx < total; idx += (long long)gridDim.x * blockDim.x) {
    const int f = (int)(idx % F);
    const float x = X[idx];
    const float* t = s_thr + f * TB;
    int code = 0;
#pragma unroll 4
    for (int j = 0; j < TB; ++j) code += (t[j] < x) ? 1 : 0;  // strict: see ops.bin_codes
    out[idx] = (uint8_t)code;
  }
}

// splitmix64-based per-(row,tree) bag weight: Poisson via inverse-CDF table
// (cdf[k] = P(X<=k)) or Bernoulli(cdf[0]=rate). Keyed by the GLOBAL row id,
// so bagging is rank-count-invariant (SURVEY.md §7 hard-part 5). The torch
// reference (ops._hash_bag_torch) computes the identical function bitwise.
__device__ __forceinline__ float flowml_hash_bag(
    int mode, const float* cdf, unsigned long long seed, long long grow, int t) {
  unsigned long long x = seed ^ (0x9E3779B97F4A7C15ULL * (unsigned long long)(grow + 1))
                              ^ (0xBF58476D1CE4E5B9ULL * (unsigned long long)(t + 1));
  x += 0x9E3779B97F4A7C15ULL;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
  x = x ^ (x >> 31);
  const float u = (float)(x >> 40) * (1.0f / 16777216.0f);  // 24-bit uniform
  if (mode == 3) return u < cdf[0] ? 1.f : 0.f;
  int k = 0;
#pragma unroll
  for (int i = 0; i < 8; ++i) k += (u >= cdf[i]) ? 1 : 0;
  return (float)k;
}

// Split routing. Continuous: go left iff code <= thr (bin space) or
// x <= threshold (raw space). Categorical (mask != 0): go left iff the
// category bit is set (CategoricalSplit.shouldGoLeft; unseen/out-of-range
// categories go right). Masks are int64 -> arity <= 63, enforced at fit.
__device__ inline bool flowml_left_code(int code, int thr, long long mask) {
  if (mask != 0) return code <= 63 && ((mask >> code) & 1LL) != 0;
  return code <= thr;
}

__device__ inline bool flowml_left_raw(float x, float thr, long long mask) {
  if (mask != 0) {
    if (x < 0.f || x > 63.f) return false;
    return ((mask >> (int)x) & 1LL) != 0;
  }
  return x <= thr;
}

// ---------------------------------------------------------------------------
// K2 — classification histogram (LDS-privatized per tree, fused advance/bag)
// ---------------------------------------------------------------------------

extern "C" __global__ void flowml_tree_hist_cls(
    const uint8_t* __restrict__ codes,   // N×F
    int* __restrict__ nodeid,            // T×N global arena ids (advance: in place)
    const int* __restrict__ lut,         // n_nodes: arena id -> chunk group, -1
    const int* __restrict__ tree_lo,     // T
    const int* __restrict__ tree_cnt,    // T (groups of tree t in chunk)
    const uint8_t* __restrict__ labels,  // N
    const float* __restrict__ bag,       // T×N or nullptr
    const int* __restrict__ fs,          // G×Fp or nullptr (identity)
    float* __restrict__ out,             // G×Fp×B×C
    const int* __restrict__ adv_feat,    // arena split arrays or null
    const int* __restrict__ adv_thr, const int* __restrict__ adv_left,
    const int* __restrict__ adv_right,
    const long long* __restrict__ adv_cat,  // arena categorical masks or null
    int bag_mode, const float* __restrict__ bag_cdf,
    unsigned long long bag_seed, long long row_offset,
    long long N, int F, int Fp, int B, int C, int l_max,
    long long rows_per_block, int stage_bytes) {
  extern __shared__ float s_hist[];
  const int t = blockIdx.x;
  const int cnt = tree_cnt[t];
  const bool advancing = adv_feat != nullptr;
  if (cnt == 0 && !advancing) return;
  // l_max == 0 ⇒ GLOBAL mode: accumulate straight into `out` with HBM
  // atomics instead of LDS privatization. Used for deep levels where the
  // frontier no longer fits LDS: one pass over the codes instead of one per
  // LDS-sized chunk. Counts are integers accumulated in fp32 (≤2^24), so
  // float atomics stay exact ⇒ bitwise deterministic in any order.
  const bool use_lds = l_max > 0;
  const int hist_sz = (cnt && use_lds) ? l_max * Fp * B * C : 0;
  for (int i = threadIdx.x; i < hist_sz; i += blockDim.x) s_hist[i] = 0.f;
  __syncthreads();

  const int g_lo = tree_lo[t];
  int* nid_t = nodeid + (long long)t * N;
  const float* bag_t = bag ? bag + (long long)t * N : nullptr;
  const long long r0 = blockIdx.y * rows_per_block;
  const long long r1 = min(r0 + rows_per_block, N);
  const int stride_fbc = Fp * B * C;
  // staged codes tile right after the histograms (16-B aligned carve):
  // per-lane byte gathers from GLOBAL split into ~F cache lines per wave
  // instruction (the TA-throughput floor); a coalesced 16-B block copy into
  // LDS followed by LDS byte reads removes that (stage_bytes==0 disables it
  // when F is too wide to fit beside the histograms).
  uint8_t* s_codes =
      (uint8_t*)(s_hist + ((l_max * (long long)stride_fbc + 3) & ~3LL));
  const bool staged = stage_bytes > 0;
  float cdf[8];
  if (bag_mode >= 2) {
#pragma unroll
    for (int i = 0; i < 8; ++i) cdf[i] = bag_cdf[i];
  }

  if (!staged) {
    // Two rows per thread, states resolved side by side: one row's
    // nid -> lut -> bag -> gathers is a serial dependent-load chain, and
    // r1 PMC showed the kernel gather-LATENCY bound (zero LDS bank
    // conflicts, 8 waves/CU) — pairing rows doubles the independent
    // chains in flight per thread.
    for (long long sub = r0; sub < r1; sub += 2LL * blockDim.x) {
      const long long rA = sub + threadIdx.x;
      const long long rB = sub + blockDim.x + threadIdx.x;
      const uint8_t* crowA = codes + rA * F;
      const uint8_t* crowB = codes + rB * F;
      int gA = -1, gB = -1, labA = 0, labB = 0;
      float wA = 1.f, wB = 1.f;
      if (rA < r1) {
        int nid = nid_t[rA];
        if (advancing && nid >= 0) {
          const int f = adv_feat[nid];
          // rows KEEP their node id at leaves (lut[leaf] = -1 already
          // skips them) so the per-leaf stats passes can address them
          if (f >= 0) {
            const long long cm = adv_cat ? adv_cat[nid] : 0LL;
            nid = flowml_left_code(crowA[f], adv_thr[nid], cm)
                      ? adv_left[nid] : adv_right[nid];
            nid_t[rA] = nid;
          }
        }
        if (cnt && nid >= 0) gA = lut[nid];
      }
      if (rB < r1) {
        int nid = nid_t[rB];
        if (advancing && nid >= 0) {
          const int f = adv_feat[nid];
          if (f >= 0) {
            const long long cm = adv_cat ? adv_cat[nid] : 0LL;
            nid = flowml_left_code(crowB[f], adv_thr[nid], cm)
                      ? adv_left[nid] : adv_right[nid];
            nid_t[rB] = nid;
          }
        }
        if (cnt && nid >= 0) gB = lut[nid];
      }
      if (bag_mode >= 2) {
        if (gA >= 0) {
          wA = flowml_hash_bag(bag_mode, cdf, bag_seed, row_offset + rA, t);
          if (wA == 0.f) gA = -1;
        }
        if (gB >= 0) {
          wB = flowml_hash_bag(bag_mode, cdf, bag_seed, row_offset + rB, t);
          if (wB == 0.f) gB = -1;
        }
      }
      if (bag_t) {
        if (gA >= 0) {
          const float bw = bag_t[rA];
          if (bw == 0.f) gA = -1; else wA *= bw;
        }
        if (gB >= 0) {
          const float bw = bag_t[rB];
          if (bw == 0.f) gB = -1; else wB *= bw;
        }
      }
      if (gA >= 0) labA = labels[rA];
      if (gB >= 0) labB = labels[rB];
      const int* fsA = (fs && gA >= 0) ? fs + (long long)gA * Fp : nullptr;
      const int* fsB = (fs && gB >= 0) ? fs + (long long)gB * Fp : nullptr;
      // branch on the address space (not a ternary pointer): keeps the LDS
      // path's atomic a ds_add_f32, not a generic flat_atomic_add_f32
      if (use_lds) {
        float* hrowA = gA >= 0 ? s_hist + (long long)(gA - g_lo) * stride_fbc
                               : nullptr;
        float* hrowB = gB >= 0 ? s_hist + (long long)(gB - g_lo) * stride_fbc
                               : nullptr;
        for (int fp = 0; fp < Fp; ++fp) {
          if (gA >= 0) {
            const int f = fsA ? fsA[fp] : fp;
            atomicAdd(&hrowA[(fp * B + crowA[f]) * C + labA], wA);
          }
          if (gB >= 0) {
            const int f = fsB ? fsB[fp] : fp;
            atomicAdd(&hrowB[(fp * B + crowB[f]) * C + labB], wB);
          }
        }
      } else {
        float* hrowA = gA >= 0 ? out + (long long)gA * stride_fbc : nullptr;
        float* hrowB = gB >= 0 ? out + (long long)gB * stride_fbc : nullptr;
        for (int fp = 0; fp < Fp; ++fp) {
          if (gA >= 0) {
            const int f = fsA ? fsA[fp] : fp;
            atomicAdd(&hrowA[(fp * B + crowA[f]) * C + labA], wA);
          }
          if (gB >= 0) {
            const int f = fsB ? fsB[fp] : fp;
            atomicAdd(&hrowB[(fp * B + crowB[f]) * C + labB], wB);
          }
        }
      }
    }
  } else
  for (long long sub = r0; sub < r1; sub += blockDim.x) {
    const int rows_in = (int)min((long long)blockDim.x, r1 - sub);
    const long long r = sub + threadIdx.x;
    __syncthreads();  // previous sub-tile fully consumed
    {
      const uint8_t* src = codes + sub * F;
      const int total = rows_in * F;
      const int total16 = total & ~15;
      for (int i = threadIdx.x * 16; i < total16; i += blockDim.x * 16)
        *reinterpret_cast<uint4*>(s_codes + i) =
            *reinterpret_cast<const uint4*>(src + i);
      for (int i = total16 + threadIdx.x; i < total; i += blockDim.x)
        s_codes[i] = src[i];
      __syncthreads();
    }
    const uint8_t* crow = s_codes + threadIdx.x * F;
    if (threadIdx.x >= rows_in) continue;
    int nid = nid_t[r];
    if (advancing) {
      if (nid >= 0) {
        const int f = adv_feat[nid];
        if (f >= 0) {
          const long long cm = adv_cat ? adv_cat[nid] : 0LL;
          nid = flowml_left_code(crow[f], adv_thr[nid], cm) ? adv_left[nid]
                                                            : adv_right[nid];
          nid_t[r] = nid;
        }
      }
      if (cnt == 0) continue;
    }
    if (nid < 0) continue;
    const int g = lut[nid];
    if (g < 0) continue;
    float w = 1.f;
    if (bag_mode >= 2) {
      w = flowml_hash_bag(bag_mode, cdf, bag_seed, row_offset + r, t);
      if (w == 0.f) continue;
    }
    if (bag_t) {
      const float bw = bag_t[r];
      if (bw == 0.f) continue;
      w *= bw;
    }
    const int lab = labels[r];
    const int* fs_g = fs ? fs + (long long)g * Fp : nullptr;
    if (use_lds) {
      float* hrow = s_hist + (long long)(g - g_lo) * stride_fbc;
      for (int fp = 0; fp < Fp; ++fp) {
        const int f = fs_g ? fs_g[fp] : fp;
        atomicAdd(&hrow[(fp * B + crow[f]) * C + lab], w);
      }
    } else {
      float* hrow = out + (long long)g * stride_fbc;
      for (int fp = 0; fp < Fp; ++fp) {
        const int f = fs_g ? fs_g[fp] : fp;
        atomicAdd(&hrow[(fp * B + crow[f]) * C + lab], w);
      }
    }
  }
  if (cnt == 0 || !use_lds) return;
  __syncthreads();
  // flush this tree's [g_lo, g_lo+cnt) groups (same layout in LDS and out)
  float* out_t = out + (long long)g_lo * stride_fbc;
  const int flush_sz = cnt * stride_fbc;
  for (int i = threadIdx.x; i < flush_sz; i += blockDim.x) {
    const float v = s_hist[i];
    if (v != 0.f) atomicAdd(&out_t[i], v);
  }
}

// Multi-tree-per-block variant (LDS mode, unstaged): TCC counters showed
// the one-tree-per-block kernel L2-REQUEST bound (96.6% hit rate, ~103G
// req/s) — every tree re-fetches each row's 41-B code line. One block
// handling `tpb` trees reuses the row line (L1-hot) across tpb visits,
// dividing the code-line request stream by ~tpb. Chosen by the launcher
// when tpb trees' chunk histograms fit the 150 KB LDS carve together.
extern "C" __global__ void flowml_tree_hist_cls_tpb(
    const uint8_t* __restrict__ codes, int* __restrict__ nodeid,
    const int* __restrict__ lut, const int* __restrict__ tree_lo,
    const int* __restrict__ tree_cnt, const uint8_t* __restrict__ labels,
    const float* __restrict__ bag, const int* __restrict__ fs,
    float* __restrict__ out, const int* __restrict__ adv_feat,
    const int* __restrict__ adv_thr, const int* __restrict__ adv_left,
    const int* __restrict__ adv_right,
    const long long* __restrict__ adv_cat,
    int bag_mode, const float* __restrict__ bag_cdf,
    unsigned long long bag_seed, long long row_offset,
    long long N, int F, int T, int tpb, int Fp, int B, int C,
    long long rows_per_block) {
  extern __shared__ float s_hist[];
  const int t0 = blockIdx.x * tpb;
  const int t_hi = min(t0 + tpb, T);
  const bool advancing = adv_feat != nullptr;
  // contiguous group span of this block's trees (tree_lo of a cnt==0 tree
  // is 0 by host convention — guard on cnt)
  int g_base = INT_MAX, g_end = 0;
  for (int tt = t0; tt < t_hi; ++tt) {
    const int c = tree_cnt[tt];
    if (c > 0) {
      g_base = min(g_base, tree_lo[tt]);
      g_end = max(g_end, tree_lo[tt] + c);
    }
  }
  const int n_groups = (g_base == INT_MAX) ? 0 : g_end - g_base;
  if (n_groups == 0 && !advancing) return;
  const int stride_fbc = Fp * B * C;
  const int hist_sz = n_groups * stride_fbc;
  for (int i = threadIdx.x; i < hist_sz; i += blockDim.x) s_hist[i] = 0.f;
  __syncthreads();
  const long long r0 = blockIdx.y * rows_per_block;
  const long long r1 = min(r0 + rows_per_block, N);
  float cdf[8];
  if (bag_mode >= 2) {
#pragma unroll
    for (int i = 0; i < 8; ++i) cdf[i] = bag_cdf[i];
  }
  for (long long r = r0 + threadIdx.x; r < r1; r += blockDim.x) {
    const uint8_t* crow = codes + r * F;
    for (int tt = t0; tt < t_hi; ++tt) {
      int* nt = nodeid + (long long)tt * N;
      int nid = nt[r];
      if (advancing) {
        if (nid >= 0) {
          const int f = adv_feat[nid];
          if (f >= 0) {  // keep-at-leaf
            const long long cm = adv_cat ? adv_cat[nid] : 0LL;
            nid = flowml_left_code(crow[f], adv_thr[nid], cm) ? adv_left[nid]
                                                              : adv_right[nid];
            nt[r] = nid;
          }
        }
        if (n_groups == 0) continue;
      }
      if (nid < 0) continue;
      const int g = lut[nid];
      if (g < 0) continue;
      float w = 1.f;
      if (bag_mode >= 2) {
        w = flowml_hash_bag(bag_mode, cdf, bag_seed, row_offset + r, tt);
        if (w == 0.f) continue;
      }
      if (bag) {
        const float bw = bag[(long long)tt * N + r];
        if (bw == 0.f) continue;
        w *= bw;
      }
      const int lab = labels[r];
      const int* fs_g = fs ? fs + (long long)g * Fp : nullptr;
      float* hrow = s_hist + (long long)(g - g_base) * stride_fbc;
      for (int fp = 0; fp < Fp; ++fp) {
        const int f = fs_g ? fs_g[fp] : fp;
        atomicAdd(&hrow[(fp * B + crow[f]) * C + lab], w);
      }
    }
  }
  if (n_groups == 0) return;
  __syncthreads();
  float* out_b = out + (long long)g_base * stride_fbc;
  for (int i = threadIdx.x; i < hist_sz; i += blockDim.x) {
    const float v = s_hist[i];
    if (v != 0.f) atomicAdd(&out_b[i], v);
  }
}

// ---------------------------------------------------------------------------
// K2b — regression histogram. Per-bin cells are (Σwt, Σw) ONLY — the
// variance split gain is algebraically free of per-bin Σt²
// (gain = (sL²/nL + sR²/nR − sP²/nP)/nP), so the third atomic per
// (row,feature) of the naive (Σt,Σt²,n) layout is replaced by ONE per-row
// atomic into a per-NODE Σwt² slot (needed only for node impurity values):
// 2·Fp+1 atomics per row instead of 3·Fp. Output layout:
//   out[0 .. G·Fp·B·2)  = cells, out[G·Fp·B·2 + g] = node Σwt².
// ---------------------------------------------------------------------------

extern "C" __global__ void flowml_tree_hist_reg(
    const uint8_t* __restrict__ codes, int* __restrict__ nodeid,
    const int* __restrict__ lut, const int* __restrict__ tree_lo,
    const int* __restrict__ tree_cnt, const float* __restrict__ targets,
    const float* __restrict__ bag, const int* __restrict__ fs,
    float* __restrict__ out,
    const int* __restrict__ adv_feat, const int* __restrict__ adv_thr,
    const int* __restrict__ adv_left, const int* __restrict__ adv_right,
    const long long* __restrict__ adv_cat,
    int bag_mode, const float* __restrict__ bag_cdf,
    unsigned long long bag_seed, long long row_offset,
    long long N, int F, int Fp, int B, int Gtot,
    int l_max, long long rows_per_block, int stage_bytes) {
  extern __shared__ float s_hist[];
  const int t = blockIdx.x;
  const int cnt = tree_cnt[t];
  const bool advancing = adv_feat != nullptr;
  if (cnt == 0 && !advancing) return;
  // l_max == 0 ⇒ GLOBAL mode (see flowml_tree_hist_cls). Regression sums
  // are floats, so atomic order shifts ulps — but the LDS path's flush
  // atomics already make it order-dependent across blocks, so GLOBAL mode
  // is the same accuracy class (parity tests are tolerance-based).
  const bool use_lds = l_max > 0;
  const int stride_fb2 = Fp * B * 2;
  const int hist_sz = (cnt && use_lds) ? l_max * (stride_fb2 + 1) : 0;
  for (int i = threadIdx.x; i < hist_sz; i += blockDim.x) s_hist[i] = 0.f;
  __syncthreads();

  const int g_lo = tree_lo[t];
  int* nid_t = nodeid + (long long)t * N;
  const float* bag_t = bag ? bag + (long long)t * N : nullptr;
  const long long r0 = blockIdx.y * rows_per_block;
  const long long r1 = min(r0 + rows_per_block, N);
  uint8_t* s_codes =
      (uint8_t*)(s_hist + ((l_max * (long long)(stride_fb2 + 1) + 3) & ~3LL));
  const bool staged = stage_bytes > 0;
  float cdf[8];
  if (bag_mode >= 2) {
#pragma unroll
    for (int i = 0; i < 8; ++i) cdf[i] = bag_cdf[i];
  }

  if (!staged) {
    // paired rows: two independent nid->lut->bag->gather chains in flight
    // per thread (same rationale and structure as the cls kernel)
    for (long long sub = r0; sub < r1; sub += 2LL * blockDim.x) {
      const long long rA = sub + threadIdx.x;
      const long long rB = sub + blockDim.x + threadIdx.x;
      const uint8_t* crowA = codes + rA * F;
      const uint8_t* crowB = codes + rB * F;
      int gA = -1, gB = -1;
      float wA = 1.f, wB = 1.f;
      if (rA < r1) {
        int nid = nid_t[rA];
        if (advancing && nid >= 0) {
          const int f = adv_feat[nid];
          if (f >= 0) {  // keep-at-leaf (see cls kernel)
            const long long cm = adv_cat ? adv_cat[nid] : 0LL;
            nid = flowml_left_code(crowA[f], adv_thr[nid], cm)
                      ? adv_left[nid] : adv_right[nid];
            nid_t[rA] = nid;
          }
        }
        if (cnt && nid >= 0) gA = lut[nid];
      }
      if (rB < r1) {
        int nid = nid_t[rB];
        if (advancing && nid >= 0) {
          const int f = adv_feat[nid];
          if (f >= 0) {
            const long long cm = adv_cat ? adv_cat[nid] : 0LL;
            nid = flowml_left_code(crowB[f], adv_thr[nid], cm)
                      ? adv_left[nid] : adv_right[nid];
            nid_t[rB] = nid;
          }
        }
        if (cnt && nid >= 0) gB = lut[nid];
      }
      if (bag_mode >= 2) {
        if (gA >= 0) {
          wA = flowml_hash_bag(bag_mode, cdf, bag_seed, row_offset + rA, t);
          if (wA == 0.f) gA = -1;
        }
        if (gB >= 0) {
          wB = flowml_hash_bag(bag_mode, cdf, bag_seed, row_offset + rB, t);
          if (wB == 0.f) gB = -1;
        }
      }
      if (bag_t) {
        if (gA >= 0) {
          const float bw = bag_t[rA];
          if (bw == 0.f) gA = -1; else wA *= bw;
        }
        if (gB >= 0) {
          const float bw = bag_t[rB];
          if (bw == 0.f) gB = -1; else wB *= bw;
        }
      }
      float wtA = 0.f, wt2A = 0.f, wtB = 0.f, wt2B = 0.f;
      if (gA >= 0) {
        const float tv = targets[rA];
        wtA = wA * tv;
        wt2A = wtA * tv;
      }
      if (gB >= 0) {
        const float tv = targets[rB];
        wtB = wB * tv;
        wt2B = wtB * tv;
      }
      const int* fsA = (fs && gA >= 0) ? fs + (long long)gA * Fp : nullptr;
      const int* fsB = (fs && gB >= 0) ? fs + (long long)gB * Fp : nullptr;
      // branch on address space: ds_add_f32 on the LDS path (see cls)
      if (use_lds) {
        float* hrowA = gA >= 0 ? s_hist + (long long)(gA - g_lo) * stride_fb2
                               : nullptr;
        float* hrowB = gB >= 0 ? s_hist + (long long)(gB - g_lo) * stride_fb2
                               : nullptr;
        for (int fp = 0; fp < Fp; ++fp) {
          if (gA >= 0) {
            const int f = fsA ? fsA[fp] : fp;
            float* cell = &hrowA[(fp * B + crowA[f]) * 2];
            atomicAdd(cell + 0, wtA);
            atomicAdd(cell + 1, wA);
          }
          if (gB >= 0) {
            const int f = fsB ? fsB[fp] : fp;
            float* cell = &hrowB[(fp * B + crowB[f]) * 2];
            atomicAdd(cell + 0, wtB);
            atomicAdd(cell + 1, wB);
          }
        }
        if (gA >= 0) atomicAdd(&s_hist[l_max * stride_fb2 + (gA - g_lo)], wt2A);
        if (gB >= 0) atomicAdd(&s_hist[l_max * stride_fb2 + (gB - g_lo)], wt2B);
      } else {
        float* hrowA = gA >= 0 ? out + (long long)gA * stride_fb2 : nullptr;
        float* hrowB = gB >= 0 ? out + (long long)gB * stride_fb2 : nullptr;
        for (int fp = 0; fp < Fp; ++fp) {
          if (gA >= 0) {
            const int f = fsA ? fsA[fp] : fp;
            float* cell = &hrowA[(fp * B + crowA[f]) * 2];
            atomicAdd(cell + 0, wtA);
            atomicAdd(cell + 1, wA);
          }
          if (gB >= 0) {
            const int f = fsB ? fsB[fp] : fp;
            float* cell = &hrowB[(fp * B + crowB[f]) * 2];
            atomicAdd(cell + 0, wtB);
            atomicAdd(cell + 1, wB);
          }
        }
        if (gA >= 0) atomicAdd(&out[(long long)Gtot * stride_fb2 + gA], wt2A);
        if (gB >= 0) atomicAdd(&out[(long long)Gtot * stride_fb2 + gB], wt2B);
      }
    }
  } else
  for (long long sub = r0; sub < r1; sub += blockDim.x) {
    const int rows_in = (int)min((long long)blockDim.x, r1 - sub);
    const long long r = sub + threadIdx.x;
    __syncthreads();
    {
      const uint8_t* src = codes + sub * F;
      const int total = rows_in * F;
      const int total16 = total & ~15;
      for (int i = threadIdx.x * 16; i < total16; i += blockDim.x * 16)
        *reinterpret_cast<uint4*>(s_codes + i) =
            *reinterpret_cast<const uint4*>(src + i);
      for (int i = total16 + threadIdx.x; i < total; i += blockDim.x)
        s_codes[i] = src[i];
      __syncthreads();
    }
    const uint8_t* crow = s_codes + threadIdx.x * F;
    if (threadIdx.x >= rows_in) continue;
    int nid = nid_t[r];
    if (advancing) {
      if (nid >= 0) {
        const int f = adv_feat[nid];
        if (f >= 0) {  // keep-at-leaf (see cls kernel)
          const long long cm = adv_cat ? adv_cat[nid] : 0LL;
          nid = flowml_left_code(crow[f], adv_thr[nid], cm) ? adv_left[nid]
                                                            : adv_right[nid];
          nid_t[r] = nid;
        }
      }
      if (cnt == 0) continue;
    }
    if (nid < 0) continue;
    const int g = lut[nid];
    if (g < 0) continue;
    float w = 1.f;
    if (bag_mode >= 2) {
      w = flowml_hash_bag(bag_mode, cdf, bag_seed, row_offset + r, t);
      if (w == 0.f) continue;
    }
    if (bag_t) {
      const float bw = bag_t[r];
      if (bw == 0.f) continue;
      w *= bw;
    }
    const float tv = targets[r];
    const float wt = w * tv;
    const float wt2 = wt * tv;
    const int* fs_g = fs ? fs + (long long)g * Fp : nullptr;
    if (use_lds) {
      const int gl = g - g_lo;
      float* hrow = s_hist + (long long)gl * stride_fb2;
      for (int fp = 0; fp < Fp; ++fp) {
        const int f = fs_g ? fs_g[fp] : fp;
        float* cell = &hrow[(fp * B + crow[f]) * 2];
        atomicAdd(cell + 0, wt);
        atomicAdd(cell + 1, w);
      }
      atomicAdd(&s_hist[l_max * stride_fb2 + gl], wt2);
    } else {
      float* hrow = out + (long long)g * stride_fb2;
      for (int fp = 0; fp < Fp; ++fp) {
        const int f = fs_g ? fs_g[fp] : fp;
        float* cell = &hrow[(fp * B + crow[f]) * 2];
        atomicAdd(cell + 0, wt);
        atomicAdd(cell + 1, w);
      }
      atomicAdd(&out[(long long)Gtot * stride_fb2 + g], wt2);
    }
  }
  if (cnt == 0 || !use_lds) return;
  __syncthreads();
  float* out_t = out + (long long)g_lo * stride_fb2;
  const int flush_sz = cnt * stride_fb2;
  for (int i = threadIdx.x; i < flush_sz; i += blockDim.x) {
    const float v = s_hist[i];
    if (v != 0.f) atomicAdd(&out_t[i], v);
  }
  float* s2_out = out + (long long)Gtot * stride_fb2 + g_lo;
  for (int i = threadIdx.x; i < cnt; i += blockDim.x) {
    const float v = s_hist[l_max * stride_fb2 + i];
    if (v != 0.f) atomicAdd(&s2_out[i], v);
  }
}

// ---------------------------------------------------------------------------
// K4b — standalone node advance (global arena ids)
// ---------------------------------------------------------------------------

extern "C" __global__ void flowml_node_advance(
    const uint8_t* __restrict__ codes, const int* __restrict__ nodeid_in,
    int* __restrict__ nodeid_out, const int* __restrict__ feat,
    const int* __restrict__ thr, const int* __restrict__ left,
    const int* __restrict__ right, const long long* __restrict__ catmask,
    long long N, int F) {
  const int t = blockIdx.y;
  const int* in_t = nodeid_in + (long long)t * N;
  int* out_t = nodeid_out + (long long)t * N;
  for (long long r = blockIdx.x * (long long)blockDim.x + threadIdx.x; r < N;
       r += (long long)gridDim.x * blockDim.x) {
    const int nid = in_t[r];
    if (nid < 0) {
      out_t[r] = nid;
      continue;
    }
    const int f = feat[nid];
    if (f < 0) {
      out_t[r] = nid;  // keep-at-leaf (rows stay addressable for stats passes)
      continue;
    }
    const int code = codes[r * (long long)F + f];
    out_t[r] = flowml_left_code(code, thr[nid], catmask[nid]) ? left[nid]
                                                              : right[nid];
  }
}

// ---------------------------------------------------------------------------
// K8 — batched ensemble inference
// ---------------------------------------------------------------------------

// Packed-node inference layout: one 16-B uint4 per node = {feature (bit 30
// flags a categorical split), threshold bits, left, right} — ONE L1/L2
// access per visit instead of four scattered SoA loads (feature/threshold/
// left/right). catmask stays a side array read only at flagged nodes.
#define FLOWML_CAT_FLAG (1 << 30)

// FLOWML_TPAR trees traverse concurrently per thread: a single tree's walk
// is a serial chain of dependent loads (feature -> compare -> child ->
// feature, ~2 loads x depth levels), so one-tree-at-a-time is latency-bound
// (measured 131G node-visits/s at 500 trees x depth 8). Interleaving
// independent walks gives the scheduler FLOWML_TPAR loads in flight per
// thread. The t-loop stays uniform across the wave, so only one tree-tile's
// node arrays are hot in L1/L2 at a time.
#define FLOWML_TPAR 4

extern "C" __global__ void flowml_tree_predict(
    const float* __restrict__ X, const uint4* __restrict__ packed,
    const float* __restrict__ value, const long long* __restrict__ catmask,
    float* __restrict__ out, long long N, int F, int T, int M, int K) {
  for (long long r = blockIdx.x * (long long)blockDim.x + threadIdx.x; r < N;
       r += (long long)gridDim.x * blockDim.x) {
    float acc[FLOWML_MAXK];
#pragma unroll
    for (int k = 0; k < FLOWML_MAXK; ++k) acc[k] = 0.f;
    const float* xr = X + r * (long long)F;
    for (int t0 = 0; t0 < T; t0 += FLOWML_TPAR) {
      const int nt = min(FLOWML_TPAR, T - t0);
      int node[FLOWML_TPAR];
      uint4 nd[FLOWML_TPAR];
#pragma unroll
      for (int j = 0; j < FLOWML_TPAR; ++j) {
        node[j] = 0;
        nd[j] = (j < nt) ? packed[(long long)(t0 + j) * M]
                         : uint4{(unsigned)-1, 0, 0, 0};
      }
      bool live = true;
      while (live) {
        live = false;
#pragma unroll
        for (int j = 0; j < FLOWML_TPAR; ++j) {
          int f = (int)nd[j].x;
          if (f < 0) continue;
          const long long base = (long long)(t0 + j) * M;
          bool go_left;
          const float thr = __uint_as_float(nd[j].y);
          if (f & FLOWML_CAT_FLAG) {
            f &= ~FLOWML_CAT_FLAG;
            const float x = xr[f];
            const long long m = catmask[base + node[j]];
            go_left = (x >= 0.f && x <= 63.f) && (((m >> (int)x) & 1LL) != 0);
          } else {
            go_left = xr[f] <= thr;
          }
          const int n = go_left ? (int)nd[j].z : (int)nd[j].w;
          node[j] = n;
          nd[j] = packed[base + n];
          live = true;
        }
      }
#pragma unroll
      for (int j = 0; j < FLOWML_TPAR; ++j) {
        if (j >= nt) continue;
        const float* v = value + ((long long)(t0 + j) * M + node[j]) * K;
        for (int k = 0; k < K; ++k) acc[k] += v[k];
      }
    }
    float* o = out + r * (long long)K;
    for (int k = 0; k < K; ++k) o[k] = acc[k];
  }
}

// ---------------------------------------------------------------------------
// K1 — fused logistic delta: logits (K dots) + softmax + delta + loss
// (gradW = deltaᵀX and gradB = Σdelta are plain GEMM/reduce — rocBLAS/torch)
// ---------------------------------------------------------------------------

extern "C" __global__ void flowml_logreg_delta(
    const float* __restrict__ X,  // N×F
    const int* __restrict__ y,    // N
    const float* __restrict__ w,  // N or nullptr
    const float* __restrict__ W,  // K×F
    const float* __restrict__ b,  // K
    float* __restrict__ delta,    // N×K out
    float* __restrict__ loss,     // 1 (atomic)
    long long N, int F, int K) {
  extern __shared__ float s_w[];  // K×F weights + K bias
  const int KF = K * F;
  for (int i = threadIdx.x; i < KF; i += blockDim.x) s_w[i] = W[i];
  float* s_b = s_w + KF;
  for (int i = threadIdx.x; i < K; i += blockDim.x) s_b[i] = b[i];
  __syncthreads();

  float loss_local = 0.f;
  for (long long r = blockIdx.x * (long long)blockDim.x + threadIdx.x; r < N;
       r += (long long)gridDim.x * blockDim.x) {
    const float* xr = X + r * (long long)F;
    float logits[FLOWML_MAXK];
    float m = -1e30f;
    for (int k = 0; k < K; ++k) {
      float acc = s_b[k];
      const float* wk = s_w + k * F;
      for (int f = 0; f < F; ++f) acc = fmaf(wk[f], xr[f], acc);
      logits[k] = acc;
      m = fmaxf(m, acc);
    }
    float sum = 0.f;
    for (int k = 0; k < K; ++k) {
      logits[k] = __expf(logits[k] - m);
      sum += logits[k];
    }
    const float inv = 1.f / sum;
    const int yk = y[r];
    const float wr = w ? w[r] : 1.f;
    // loss = logZ - logit_y = log(sum) - (logit_y - m) = log(sum) - log(e_y)
    loss_local += wr * (logf(sum) - logf(logits[yk]));
    float* dr = delta + r * (long long)K;
    for (int k = 0; k < K; ++k) {
      float p = logits[k] * inv;
      dr[k] = wr * (p - (k == yk ? 1.f : 0.f));
    }
  }
  // block-reduce loss
  __shared__ float s_red[256];
  s_red[threadIdx.x] = loss_local;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) s_red[threadIdx.x] += s_red[threadIdx.x + s];
    __syncthreads();
  }
  // per-block partial (fixed-order host sum ⇒ deterministic loss)
  if (threadIdx.x == 0) loss[blockIdx.x] = s_red[0];
}

// ---------------------------------------------------------------------------
// K6 — softmax / cross-entropy: loss + delta from logits
// ---------------------------------------------------------------------------

extern "C" __global__ void flowml_softmax_xent(
    const float* __restrict__ logits,  // N×K
    const int* __restrict__ y,         // N
    const float* __restrict__ w,       // N or nullptr
    float* __restrict__ delta,         // N×K out
    float* __restrict__ loss,          // 1 (atomic)
    long long N, int K) {
  float loss_local = 0.f;
  for (long long r = blockIdx.x * (long long)blockDim.x + threadIdx.x; r < N;
       r += (long long)gridDim.x * blockDim.x) {
    const float* lr = logits + r * (long long)K;
    float m = -1e30f;
    for (int k = 0; k < K; ++k) m = fmaxf(m, lr[k]);
    float sum = 0.f;
    float e[FLOWML_MAXK];
    for (int k = 0; k < K; ++k) {
      e[k] = __expf(lr[k] - m);
      sum += e[k];
    }
    const float inv = 1.f / sum;
    const int yk = y[r];
    const float wr = w ? w[r] : 1.f;
    loss_local += wr * (logf(sum) + m - lr[yk]);
    float* dr = delta + r * (long long)K;
    for (int k = 0; k < K; ++k)
      dr[k] = wr * (e[k] * inv - (k == yk ? 1.f : 0.f));
  }
  __shared__ float s_red[256];
  s_red[threadIdx.x] = loss_local;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) s_red[threadIdx.x] += s_red[threadIdx.x + s];
    __syncthreads();
  }
  // per-block partial (fixed-order host sum ⇒ deterministic loss)
  if (threadIdx.x == 0) loss[blockIdx.x] = s_red[0];
}

// ---------------------------------------------------------------------------
// Host launchers (called from ext.cpp; stream passed as void* to keep the
// binding TU free of HIP headers)
// ---------------------------------------------------------------------------

static inline int grid_1d(long long total, int block, int cap = 4096) {
  long long g = (total + block - 1) / block;
  return (int)(g < cap ? (g > 0 ? g : 1) : cap);
}

extern "C" void launch_bin_codes(const float* X, const float* thr, uint8_t* out,
                                 long long N, int F, int TB, void* stream) {
  const int block = 256;
  const size_t lds = (size_t)F * TB * sizeof(float);
  hipLaunchKernelGGL(flowml_bin_codes, dim3(grid_1d(N * F, block)), dim3(block),
                     lds, (hipStream_t)stream, X, thr, out, N, F, TB);
}

static bool g_hist_lds_opted = false;
static void hist_opt_in_big_lds() {
  if (g_hist_lds_opted) return;
  // gfx950 allows >64 KiB LDS per workgroup; opt in once per process.
  (void)hipFuncSetAttribute((const void*)flowml_tree_hist_cls,
                            hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
  (void)hipFuncSetAttribute((const void*)flowml_tree_hist_reg,
                            hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
  g_hist_lds_opted = true;
}

extern "C" void launch_tree_hist_cls(
    const uint8_t* codes, int* nodeid, const int* lut, const int* tree_lo,
    const int* tree_cnt, const uint8_t* labels, const float* bag, const int* fs,
    float* out, const int* adv_feat, const int* adv_thr, const int* adv_left,
    const int* adv_right, const long long* adv_cat, int bag_mode,
    const float* bag_cdf, unsigned long long bag_seed, long long row_offset,
    long long N, int F, int T, int Fp, int B, int C,
    int l_max, void* stream) {
  hist_opt_in_big_lds();
  static const int cfg_block = flowml_env_int("FLOWML_HIST_BLOCK", 512);
  static const int cfg_rpb = flowml_env_int("FLOWML_HIST_RPB", 16384);
  const int block = cfg_block;
  // target >= 1024 blocks so a single-tree launch (GBT) still fills the
  // 256-CU chip; large T (RF) keeps the big tile for flush amortization
  long long rows_per_block = cfg_rpb;
  long long ideal = (N * (long long)T) / 1024;  // tiles*T ~ 1024 blocks
  if (ideal < 2048) ideal = 2048;
  if (ideal < rows_per_block) rows_per_block = ideal;
  const int tiles = (int)((N + rows_per_block - 1) / rows_per_block);
  const size_t hist_b = ((size_t)l_max * Fp * B * C * sizeof(float) + 15) & ~15ULL;
  // measured: LDS-staged code tiles LOSE ~25% here — adjacent lanes' 41-B
  // rows share 64-B cache lines, so L1 already coalesces the byte gathers
  // and the staging barriers/occupancy cost more than they save. Off by
  // default; keep for wider-F workloads via FLOWML_HIST_STAGE=1.
  static const int cfg_stage = flowml_env_int("FLOWML_HIST_STAGE", 0);
  int stage_bytes = cfg_stage ? block * F : 0;
  if (hist_b + (size_t)stage_bytes > 150 * 1024) stage_bytes = 0;  // fall back
  // multi-tree blocks when several trees' chunk hists share the LDS carve:
  // reuses each row's code line across tpb tree visits (L2-request bound
  // kernel — see profiles/pmc_r02_summary.md)
  // measured WORSE at default (levels 0-2: 2.85 -> 3.1-3.4 ms): the 96.6%
  // L2 hit rate already amortizes cross-tree code-line reuse, and the
  // serial per-row tree loop adds more latency than the saved requests.
  // Kept behind the env flag as a documented negative result.
  static const int cfg_tpb = flowml_env_int("FLOWML_HIST_TPB", 1);
  int tpb = 1;
  if (l_max > 0 && stage_bytes == 0 && hist_b > 0 && cfg_tpb > 1) {
    // keep 2 blocks/CU resident (<=75 KB/block): halving occupancy for
    // tpb=2 at the deepest level measured worse than the traffic saved
    tpb = (int)((75 * 1024) / hist_b);
    if (tpb > cfg_tpb) tpb = cfg_tpb;
    if (tpb > T) tpb = T;
    if (tpb < 1) tpb = 1;
  }
  if (tpb > 1) {
    const int xblocks = (T + tpb - 1) / tpb;
    const size_t lds2 = hist_b * tpb;
    hipLaunchKernelGGL(flowml_tree_hist_cls_tpb, dim3(xblocks, tiles),
                       dim3(block), lds2, (hipStream_t)stream, codes, nodeid,
                       lut, tree_lo, tree_cnt, labels, bag, fs, out, adv_feat,
                       adv_thr, adv_left, adv_right, adv_cat, bag_mode,
                       bag_cdf, bag_seed, row_offset, N, F, T, tpb, Fp, B, C,
                       rows_per_block);
    return;
  }
  const size_t lds = hist_b + stage_bytes;
  hipLaunchKernelGGL(flowml_tree_hist_cls, dim3(T, tiles), dim3(block), lds,
                     (hipStream_t)stream, codes, nodeid, lut, tree_lo, tree_cnt,
                     labels, bag, fs, out, adv_feat, adv_thr, adv_left,
                     adv_right, adv_cat, bag_mode, bag_cdf, bag_seed,
                     row_offset, N, F, Fp, B, C, l_max, rows_per_block,
                     stage_bytes);
}

extern "C" void launch_tree_hist_reg(
    const uint8_t* codes, int* nodeid, const int* lut, const int* tree_lo,
    const int* tree_cnt, const float* targets, const float* bag, const int* fs,
    float* out, const int* adv_feat, const int* adv_thr, const int* adv_left,
    const int* adv_right, const long long* adv_cat, int bag_mode,
    const float* bag_cdf, unsigned long long bag_seed, long long row_offset,
    long long N, int F, int T, int Fp, int B, int Gtot, int l_max,
    void* stream) {
  hist_opt_in_big_lds();
  static const int cfg_block = flowml_env_int("FLOWML_HIST_BLOCK", 512);
  static const int cfg_rpb = flowml_env_int("FLOWML_HIST_RPB", 16384);
  const int block = cfg_block;
  // target >= 1024 blocks so a single-tree launch (GBT) still fills the
  // 256-CU chip; large T (RF) keeps the big tile for flush amortization
  long long rows_per_block = cfg_rpb;
  long long ideal = (N * (long long)T) / 1024;  // tiles*T ~ 1024 blocks
  if (ideal < 2048) ideal = 2048;
  if (ideal < rows_per_block) rows_per_block = ideal;
  const int tiles = (int)((N + rows_per_block - 1) / rows_per_block);
  // cells (Fp*B*2 per node) + one node-s2 slot per node
  const size_t hist_b =
      ((size_t)l_max * (Fp * B * 2 + 1) * sizeof(float) + 15) & ~15ULL;
  static const int cfg_stage = flowml_env_int("FLOWML_HIST_STAGE", 0);
  int stage_bytes = cfg_stage ? block * F : 0;
  if (hist_b + (size_t)stage_bytes > 150 * 1024) stage_bytes = 0;  // fall back
  const size_t lds = hist_b + stage_bytes;
  hipLaunchKernelGGL(flowml_tree_hist_reg, dim3(T, tiles), dim3(block), lds,
                     (hipStream_t)stream, codes, nodeid, lut, tree_lo, tree_cnt,
                     targets, bag, fs, out, adv_feat, adv_thr, adv_left,
                     adv_right, adv_cat, bag_mode, bag_cdf, bag_seed,
                     row_offset, N, F, Fp, B, Gtot, l_max, rows_per_block,
                     stage_bytes);
}

extern "C" void launch_node_advance(const uint8_t* codes, const int* nodeid_in,
                                    int* nodeid_out, const int* feat,
                                    const int* thr, const int* left,
                                    const int* right, const long long* catmask,
                                    long long N, int F, int T, void* stream) {
  const int block = 256;
  hipLaunchKernelGGL(flowml_node_advance, dim3(grid_1d(N, block, 2048), T),
                     dim3(block), 0, (hipStream_t)stream, codes, nodeid_in,
                     nodeid_out, feat, thr, left, right, catmask, N, F);
}

extern "C" void launch_tree_predict(const float* X, const void* packed,
                                    const float* value,
                                    const long long* catmask, float* out,
                                    long long N, int F, int T, int M, int K,
                                    void* stream) {
  const int block = 256;
  hipLaunchKernelGGL(flowml_tree_predict, dim3(grid_1d(N, block, 8192)),
                     dim3(block), 0, (hipStream_t)stream, X,
                     (const uint4*)packed, value, catmask, out, N, F, T, M, K);
}

extern "C" void launch_logreg_delta(const float* X, const int* y,
                                    const float* w, const float* W,
                                    const float* b, float* delta, float* loss,
                                    long long N, int F, int K, void* stream) {
  const int block = 256;
  const size_t lds = (size_t)(K * F + K) * sizeof(float);
  hipLaunchKernelGGL(flowml_logreg_delta, dim3(grid_1d(N, block, 2048)),
                     dim3(block), lds, (hipStream_t)stream, X, y, w, W, b,
                     delta, loss, N, F, K);
}

extern "C" int softmax_xent_grid(long long N) {
  return grid_1d(N, 256, 2048);
}

extern "C" void launch_softmax_xent(const float* logits, const int* y,
                                    const float* w, float* delta, float* loss,
                                    long long N, int K, void* stream) {
  const int block = 256;
  hipLaunchKernelGGL(flowml_softmax_xent, dim3(softmax_xent_grid(N)),
                     dim3(block), 0, (hipStream_t)stream, logits, y, w, delta,
                     loss, N, K);
}

// ---------------------------------------------------------------------------
// K3 — split-gain scan: per (node, feature) prefix sums over bins, impurity
// gain, argmax over (feature, bin). One workgroup per node; the node's
// histogram slab is staged in LDS; gains are computed in DOUBLE with exactly
// the torch reference's formula order (flowml.classification.tree.split_scan)
// so the chosen splits match the CPU path bitwise (counts are integer-valued
// f32 -> exact in double). Output per node: packed row
// [best_f, best_bin, gain, imp_l, imp_r, cnt_l, cnt_r, left_stats[S], right_stats[S]].
// ---------------------------------------------------------------------------

__device__ __forceinline__ double flowml_impurity(const double* s, int S, int code) {
#pragma clang fp contract(off)
  // NOTE: operation ORDER mirrors the torch reference exactly (sum the
  // series first, then the final subtract/negate) so gains match bitwise
  // and argmax tie-breaks agree with the CPU path.
  if (code <= 1) {  // 0 = gini, 1 = entropy (class counts)
    double n = 0;
    for (int i = 0; i < S; ++i) n += s[i];
    if (n < 1e-12) n = 1e-12;
    if (code == 0) {
      double acc = 0.0;
      for (int i = 0; i < S; ++i) {
        const double p = s[i] / n;
        acc += p * p;
      }
      return 1.0 - acc;
    }
    double acc = 0.0;
    for (int i = 0; i < S; ++i) {
      const double p = s[i] / n;
      if (p > 0) {
        const double pc = p < 1e-30 ? 1e-30 : p;
        acc += p * log2(pc);
      }
    }
    return -acc;
  }
  // 2 = variance on (sum, sum2, n)
  double nn = s[2] < 1e-12 ? 1e-12 : s[2];
  const double m = s[0] / nn;
  double v = s[1] / nn - m * m;
  return v < 0 ? 0 : v;
}

__device__ __forceinline__ double flowml_count_of(const double* s, int S, int code) {
#pragma clang fp contract(off)
  if (code == 2) return s[2];
  double n = 0;
  for (int i = 0; i < S; ++i) n += s[i];
  return n;
}

#define K3_MAXS 8

extern "C" __global__ void flowml_split_scan(
    const float* __restrict__ hist,  // G×Fp×B×S
    float* __restrict__ packed,      // G×(7+2S)
    int G, int Fp, int B, int S, int imp_code, int min_instances,
    double min_info_gain) {
#pragma clang fp contract(off)
  extern __shared__ float s_h[];  // Fp×B×S staged slab
  const int g = blockIdx.x;
  const int slab = Fp * B * S;
  const float* hg = hist + (long long)g * slab;
  for (int i = threadIdx.x; i < slab; i += blockDim.x) s_h[i] = hg[i];
  __syncthreads();

  // parent stats from feature 0 (identical across features)
  double parent[K3_MAXS];
  for (int i = 0; i < S; ++i) {
    double acc = 0;
    for (int b = 0; b < B; ++b) acc += (double)s_h[(0 * B + b) * S + i];
    parent[i] = acc;
  }
  const double n_parent_raw = flowml_count_of(parent, S, imp_code);
  const double n_parent = n_parent_raw < 1e-12 ? 1e-12 : n_parent_raw;
  const double imp_parent = flowml_impurity(parent, S, imp_code);

  // each thread scans a subset of features; prefix over bins is sequential
  // per feature (B=32: trivially cheap from LDS)
  double best_gain = -1.0 / 0.0;
  int best_idx = 0x7FFFFFFF;  // (f*(B-1)+b); smaller wins ties (torch argmax)
  for (int f = threadIdx.x; f < Fp; f += blockDim.x) {
    // per-feature totals (torch subtracts the FEATURE's own cumsum total)
    double total_f[K3_MAXS];
    for (int i = 0; i < S; ++i) {
      double acc = 0;
      for (int b = 0; b < B; ++b) acc += (double)s_h[(f * B + b) * S + i];
      total_f[i] = acc;
    }
    double L[K3_MAXS];
    for (int i = 0; i < S; ++i) L[i] = 0;
    for (int b = 0; b < B - 1; ++b) {
      for (int i = 0; i < S; ++i) L[i] += (double)s_h[(f * B + b) * S + i];
      double R[K3_MAXS];
      for (int i = 0; i < S; ++i) R[i] = total_f[i] - L[i];
      const double nL = flowml_count_of(L, S, imp_code);
      const double nR = flowml_count_of(R, S, imp_code);
      if (nL < min_instances || nR < min_instances) continue;
      const double child =
          (nL * flowml_impurity(L, S, imp_code) + nR * flowml_impurity(R, S, imp_code)) /
          n_parent;
      const double gain = imp_parent - child;
      const int idx = f * (B - 1) + b;
      if (gain > best_gain || (gain == best_gain && idx < best_idx)) {
        best_gain = gain;
        best_idx = idx;
      }
    }
  }

  // block argmax (gain desc, idx asc)
  __shared__ double r_gain[256];
  __shared__ int r_idx[256];
  r_gain[threadIdx.x] = best_gain;
  r_idx[threadIdx.x] = best_idx;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) {
      const double og = r_gain[threadIdx.x + s];
      const int oi = r_idx[threadIdx.x + s];
      if (og > r_gain[threadIdx.x] ||
          (og == r_gain[threadIdx.x] && oi < r_idx[threadIdx.x])) {
        r_gain[threadIdx.x] = og;
        r_idx[threadIdx.x] = oi;
      }
    }
    __syncthreads();
  }

  if (threadIdx.x == 0) {
    double gain = r_gain[0];
    int idx = r_idx[0];
    float* out = packed + (long long)g * (8 + 3 * S);
    // trailing columns: the node's OWN impurity + stats (lets the grower
    // skip the dedicated root pass — roots get stats from their own scan)
    out[7 + 2 * S] = (float)imp_parent;
    for (int i = 0; i < S; ++i) out[8 + 2 * S + i] = (float)parent[i];
    const bool ok = (idx != 0x7FFFFFFF) && (gain > 0) && (gain >= min_info_gain) &&
                    !isinf(gain) && !isnan(gain);
    if (!ok) {
      out[0] = -1.f;
      for (int i = 1; i < 7 + 2 * S; ++i) out[i] = 0.f;
      return;
    }
    const int f = idx / (B - 1);
    const int b = idx % (B - 1);
    double L[K3_MAXS], R[K3_MAXS], total_f[K3_MAXS];
    for (int i = 0; i < S; ++i) {
      double acc = 0;
      for (int bb = 0; bb < B; ++bb) acc += (double)s_h[(f * B + bb) * S + i];
      total_f[i] = acc;
    }
    for (int i = 0; i < S; ++i) L[i] = 0;
    for (int bb = 0; bb <= b; ++bb)
      for (int i = 0; i < S; ++i) L[i] += (double)s_h[(f * B + bb) * S + i];
    for (int i = 0; i < S; ++i) R[i] = total_f[i] - L[i];
    out[0] = (float)f;
    out[1] = (float)b;
    out[2] = (float)gain;
    out[3] = (float)flowml_impurity(L, S, imp_code);
    out[4] = (float)flowml_impurity(R, S, imp_code);
    out[5] = (float)flowml_count_of(L, S, imp_code);
    out[6] = (float)flowml_count_of(R, S, imp_code);
    for (int i = 0; i < S; ++i) out[7 + i] = (float)L[i];
    for (int i = 0; i < S; ++i) out[7 + S + i] = (float)R[i];
  }
}

extern "C" void launch_split_scan(const float* hist, float* packed, int G,
                                  int Fp, int B, int S, int imp_code,
                                  int min_instances, double min_info_gain,
                                  void* stream) {
  const int block = 256;
  const size_t lds = (size_t)Fp * B * S * sizeof(float);
  hipLaunchKernelGGL(flowml_split_scan, dim3(G), dim3(block), lds,
                     (hipStream_t)stream, hist, packed, G, Fp, B, S, imp_code,
                     min_instances, min_info_gain);
}

// ---------------------------------------------------------------------------
// K3b — variance split scan on the 2-wide regression layout
// (cells (Σwt,Σw) + per-node Σwt²; see K2b). Canonical gain — identical op
// order to ops.split_scan_packed_reg so CPU/GPU splits match bitwise:
//   gain = (sL·sL/nL + sR·sR/nR − sP·sP/nP) / nP
// Packed per-node row (10 f32): [f, b, gain, sL, nL, sR, nR, imp_self,
// s_self, n_self]; no-split rows set f = −1 but STILL carry imp_self/s/n.
// ---------------------------------------------------------------------------

extern "C" __global__ void flowml_split_scan_reg(
    const float* __restrict__ hist,  // G·Fp·B·2 cells then G node-s2
    float* __restrict__ packed,      // G×10
    int G, int Fp, int B, int min_instances, double min_info_gain) {
#pragma clang fp contract(off)
  extern __shared__ float s_h[];  // Fp×B×2 staged slab
  const int g = blockIdx.x;
  const int slab = Fp * B * 2;
  const float* hg = hist + (long long)g * slab;
  for (int i = threadIdx.x; i < slab; i += blockDim.x) s_h[i] = hg[i];
  __syncthreads();

  // parent totals from feature 0 (ascending-b order = torch cumsum order)
  double sP = 0, nP_raw = 0;
  for (int b = 0; b < B; ++b) {
    sP += (double)s_h[(0 * B + b) * 2 + 0];
    nP_raw += (double)s_h[(0 * B + b) * 2 + 1];
  }
  const double nP = nP_raw < 1e-12 ? 1e-12 : nP_raw;
  const double s2P = (double)hist[(long long)G * slab + g];
  const double m = sP / nP;
  double imp_self = s2P / nP - m * m;
  if (imp_self < 0) imp_self = 0;
  const double gP = sP * sP / nP;

  double best_gain = -1.0 / 0.0;
  int best_idx = 0x7FFFFFFF;
  for (int f = threadIdx.x; f < Fp; f += blockDim.x) {
    double totS = 0, totN = 0;
    for (int b = 0; b < B; ++b) {
      totS += (double)s_h[(f * B + b) * 2 + 0];
      totN += (double)s_h[(f * B + b) * 2 + 1];
    }
    double sL = 0, nL = 0;
    for (int b = 0; b < B - 1; ++b) {
      sL += (double)s_h[(f * B + b) * 2 + 0];
      nL += (double)s_h[(f * B + b) * 2 + 1];
      const double sR = totS - sL;
      const double nR = totN - nL;
      if (nL < min_instances || nR < min_instances) continue;
      const double gL = sL * sL / nL;
      const double gR = sR * sR / nR;
      const double gain = (gL + gR - gP) / nP;
      const int idx = f * (B - 1) + b;
      if (gain > best_gain || (gain == best_gain && idx < best_idx)) {
        best_gain = gain;
        best_idx = idx;
      }
    }
  }

  __shared__ double r_gain[256];
  __shared__ int r_idx[256];
  r_gain[threadIdx.x] = best_gain;
  r_idx[threadIdx.x] = best_idx;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) {
      const double og = r_gain[threadIdx.x + s];
      const int oi = r_idx[threadIdx.x + s];
      if (og > r_gain[threadIdx.x] ||
          (og == r_gain[threadIdx.x] && oi < r_idx[threadIdx.x])) {
        r_gain[threadIdx.x] = og;
        r_idx[threadIdx.x] = oi;
      }
    }
    __syncthreads();
  }

  if (threadIdx.x == 0) {
    const double gain = r_gain[0];
    const int idx = r_idx[0];
    float* out = packed + (long long)g * 10;
    const bool ok = (idx != 0x7FFFFFFF) && (gain > 0) &&
                    (gain >= min_info_gain) && !isinf(gain) && !isnan(gain);
    out[7] = (float)imp_self;
    out[8] = (float)sP;
    out[9] = (float)nP_raw;
    if (!ok) {
      out[0] = -1.f;
      for (int i = 1; i < 7; ++i) out[i] = 0.f;
      return;
    }
    const int f = idx / (B - 1);
    const int b = idx % (B - 1);
    double totS = 0, totN = 0;
    for (int bb = 0; bb < B; ++bb) {
      totS += (double)s_h[(f * B + bb) * 2 + 0];
      totN += (double)s_h[(f * B + bb) * 2 + 1];
    }
    double sL = 0, nL = 0;
    for (int bb = 0; bb <= b; ++bb) {
      sL += (double)s_h[(f * B + bb) * 2 + 0];
      nL += (double)s_h[(f * B + bb) * 2 + 1];
    }
    out[0] = (float)f;
    out[1] = (float)b;
    out[2] = (float)gain;
    out[3] = (float)sL;
    out[4] = (float)nL;
    out[5] = (float)(totS - sL);
    out[6] = (float)(totN - nL);
  }
}

extern "C" void launch_split_scan_reg(const float* hist, float* packed, int G,
                                      int Fp, int B, int min_instances,
                                      double min_info_gain, void* stream) {
  const int block = 256;
  const size_t lds = (size_t)Fp * B * 2 * sizeof(float);
  hipLaunchKernelGGL(flowml_split_scan_reg, dim3(G), dim3(block), lds,
                     (hipStream_t)stream, hist, packed, G, Fp, B,
                     min_instances, min_info_gain);
}
