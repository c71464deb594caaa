// CDNA4 (gfx950) fused GEMM kernels for the MLP stack.
//
// out[N,M] = act(X[N,K] @ W[M,K]^T + b[M])
//
// Shapes are tall-skinny (N up to the 1M-flow shard, M ∈ {128,64,5},
// K ∈ {41,128,64} — BASELINE.json:8), so the design is: one 256-thread block
// = 4 waves covers 64 rows × all M columns; each wave owns a 16-row stripe
// and accumulates ceil(M/16) MFMA fragments across the K loop. X fragments
// are 8 contiguous bf16 per lane straight from HBM (coalesced across the
// 16-lane row groups); W is tiny (≤32 KiB) and L1/L2-resident, so no LDS
// staging is needed — the kernel is X-bandwidth-bound, which is the roofline
// for these shapes. K must be padded to a multiple of 32 (bf16) / 4 (f32) by
// the caller (flowml/kernels/ext.cpp pads).
//
// MFMA: mfma_f32_16x16x32_bf16 (bf16) / mfma_f32_16x16x4f32 (exact f32).
// Fragment maps (cdna_hip_programming.md §3): A[i=l%16][k=(l/16)*8+j],
// B[k=(l/16)*8+j][n=l%16], D col=l%16, row=(l/16)*4+reg.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>
#include <stdlib.h>

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define MAX_NFRAG 8  // supports M up to 128 per kernel pass

__device__ __forceinline__ float act_apply(float v, int act) {
  if (act == 1) return v > 0.f ? v : 0.f;          // relu
  if (act == 2) return 1.f / (1.f + __expf(-v));   // sigmoid
  return v;
}

// ---------------------------------------------------------------------------
// bf16 path: X[N,Kp] bf16 (Kp%32==0), W[M,Kp] bf16, out[N,M] bf16
// ---------------------------------------------------------------------------

extern "C" __global__ __launch_bounds__(256) void flowml_gemm_bias_act_bf16(
    const __bf16* __restrict__ X, const __bf16* __restrict__ W,
    const float* __restrict__ b, __bf16* __restrict__ out, long long N, int M,
    int Kp, int act) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const long long row_base = (long long)blockIdx.x * 64 + wave * 16;
  if (row_base >= N) return;
  const int i_in_tile = lane & 15;   // A row within the 16-row stripe
  const int kgrp = lane >> 4;        // 0..3 -> k offset (kgrp*8)
  const int nfrag = (M + 15) / 16;

  f32x4 acc[MAX_NFRAG];
#pragma unroll
  for (int nf = 0; nf < MAX_NFRAG; ++nf) acc[nf] = {0.f, 0.f, 0.f, 0.f};

  const long long a_row = row_base + i_in_tile;
  const bool a_valid = a_row < N;
  const __bf16* a_base = X + a_row * Kp + kgrp * 8;

  for (int kk = 0; kk < Kp; kk += 32) {
    bf16x8 a_frag;
    if (a_valid) {
      a_frag = *reinterpret_cast<const bf16x8*>(a_base + kk);
    } else {
      a_frag = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
    for (int nf = 0; nf < nfrag; ++nf) {
      const int n = nf * 16 + i_in_tile;
      bf16x8 b_frag;
      if (n < M) {
        b_frag = *reinterpret_cast<const bf16x8*>(W + (long long)n * Kp + kk + kgrp * 8);
      } else {
        b_frag = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      }
      acc[nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc[nf], 0, 0, 0);
    }
  }

  // epilogue: D col = lane&15, row = (lane>>4)*4 + reg
  const int col_in = lane & 15;
  const int row_off = (lane >> 4) * 4;
  for (int nf = 0; nf < nfrag; ++nf) {
    const int n = nf * 16 + col_in;
    if (n >= M) continue;
    const float bias = b[n];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const long long r = row_base + row_off + reg;
      if (r < N) {
        out[r * M + n] = (__bf16)act_apply(acc[nf][reg] + bias, act);
      }
    }
  }
}

// dgrad with fused activation-backward: out[N,M] = (D[N,Kp] @ Wt[M,Kp]^T)
// * act'(H[N,M]).  Wt is the layer weight TRANSPOSED once per step (tiny);
// H is the layer's forward activation. Replaces the rocBLAS dgrad GEMM +
// the separate act_backward pass (one write of delta instead of
// write+read+write — the two act_backward launches were 0.28 ms of the
// 1.47 ms step).
extern "C" __global__ __launch_bounds__(256) void flowml_gemm_dgrad_act_bf16(
    const __bf16* __restrict__ D, const __bf16* __restrict__ Wt,
    const __bf16* __restrict__ H, __bf16* __restrict__ out, long long N,
    int M, int Kp, int act) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const long long row_base = (long long)blockIdx.x * 64 + wave * 16;
  if (row_base >= N) return;
  const int i_in_tile = lane & 15;
  const int kgrp = lane >> 4;
  const int nfrag = (M + 15) / 16;

  f32x4 acc[MAX_NFRAG];
#pragma unroll
  for (int nf = 0; nf < MAX_NFRAG; ++nf) acc[nf] = {0.f, 0.f, 0.f, 0.f};

  const long long a_row = row_base + i_in_tile;
  const bool a_valid = a_row < N;
  const __bf16* a_base = D + a_row * Kp + kgrp * 8;

  for (int kk = 0; kk < Kp; kk += 32) {
    bf16x8 a_frag;
    if (a_valid) {
      a_frag = *reinterpret_cast<const bf16x8*>(a_base + kk);
    } else {
      a_frag = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
    for (int nf = 0; nf < nfrag; ++nf) {
      const int n = nf * 16 + i_in_tile;
      bf16x8 b_frag;
      if (n < M) {
        b_frag = *reinterpret_cast<const bf16x8*>(Wt + (long long)n * Kp + kk + kgrp * 8);
      } else {
        b_frag = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      }
      acc[nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc[nf], 0, 0, 0);
    }
  }

  const int col_in = lane & 15;
  const int row_off = (lane >> 4) * 4;
  for (int nf = 0; nf < nfrag; ++nf) {
    const int n = nf * 16 + col_in;
    if (n >= M) continue;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const long long r = row_base + row_off + reg;
      if (r < N) {
        const float h = (float)H[r * M + n];
        float g;
        if (act == 1) {
          g = h > 0.f ? acc[nf][reg] : 0.f;            // relu'
        } else if (act == 2) {
          g = acc[nf][reg] * h * (1.f - h);            // sigmoid'
        } else {
          g = acc[nf][reg];
        }
        out[r * M + n] = (__bf16)g;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// f32 path: exact f32 MFMA (v_mfma_f32_16x16x4_f32). Kp%4==0.
// A lane map: A[i=l%16][k=l/16]; B[k=l/16][n=l%16]; D as above.
// ---------------------------------------------------------------------------

extern "C" __global__ __launch_bounds__(256) void flowml_gemm_bias_act_f32(
    const float* __restrict__ X, const float* __restrict__ W,
    const float* __restrict__ b, float* __restrict__ out, long long N, int M,
    int Kp, int act) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const long long row_base = (long long)blockIdx.x * 64 + wave * 16;
  if (row_base >= N) return;
  const int i_in_tile = lane & 15;
  const int kgrp = lane >> 4;  // 0..3 -> k index within the 4-deep step
  const int nfrag = (M + 15) / 16;

  f32x4 acc[MAX_NFRAG];
#pragma unroll
  for (int nf = 0; nf < MAX_NFRAG; ++nf) acc[nf] = {0.f, 0.f, 0.f, 0.f};

  const long long a_row = row_base + i_in_tile;
  const bool a_valid = a_row < N;
  const float* a_base = X + a_row * Kp + kgrp;

  for (int kk = 0; kk < Kp; kk += 4) {
    const float a_elem = a_valid ? a_base[kk] : 0.f;
    for (int nf = 0; nf < nfrag; ++nf) {
      const int n = nf * 16 + i_in_tile;
      const float b_elem = (n < M) ? W[(long long)n * Kp + kk + kgrp] : 0.f;
      acc[nf] = __builtin_amdgcn_mfma_f32_16x16x4f32(a_elem, b_elem, acc[nf], 0, 0, 0);
    }
  }

  const int col_in = lane & 15;
  const int row_off = (lane >> 4) * 4;
  for (int nf = 0; nf < nfrag; ++nf) {
    const int n = nf * 16 + col_in;
    if (n >= M) continue;
    const float bias = b[n];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const long long r = row_base + row_off + reg;
      if (r < N) {
        out[r * M + n] = act_apply(acc[nf][reg] + bias, act);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// elementwise activation-backward: delta *= act'(h) (K7 epilogue fusion)
// relu: h>0 ? 1 : 0 ; sigmoid: h*(1-h) — h is the post-activation value.
// ---------------------------------------------------------------------------



static int flowml_env_int_mlp(const char* name, int dflt) {
  const char* v = getenv(name);
  return v && *v ? atoi(v) : dflt;
}

// ---------------------------------------------------------------------------
// Host launchers
// ---------------------------------------------------------------------------

extern "C" void launch_gemm_bias_act_bf16(const void* X, const void* W,
                                          const float* b, void* out,
                                          long long N, int M, int Kp, int act,
                                          void* stream) {
  const int blocks = (int)((N + 63) / 64);
  hipLaunchKernelGGL(flowml_gemm_bias_act_bf16, dim3(blocks), dim3(256), 0,
                     (hipStream_t)stream, (const __bf16*)X, (const __bf16*)W, b,
                     (__bf16*)out, N, M, Kp, act);
}

extern "C" void launch_gemm_bias_act_f32(const float* X, const float* W,
                                         const float* b, float* out,
                                         long long N, int M, int Kp, int act,
                                         void* stream) {
  const int blocks = (int)((N + 63) / 64);
  hipLaunchKernelGGL(flowml_gemm_bias_act_f32, dim3(blocks), dim3(256), 0,
                     (hipStream_t)stream, X, W, b, out, N, M, Kp, act);
}

extern "C" void launch_gemm_dgrad_act_bf16(const void* D, const void* Wt,
                                           const void* H, void* out,
                                           long long N, int M, int Kp,
                                           int act, void* stream) {
  const long long blocks = (N + 63) / 64;
  hipLaunchKernelGGL(flowml_gemm_dgrad_act_bf16, dim3((unsigned)blocks),
                     dim3(256), 0, (hipStream_t)stream, (const __bf16*)D,
                     (const __bf16*)Wt, (const __bf16*)H, (__bf16*)out, N, M,
                     Kp, act);
}


// ---------------------------------------------------------------------------
// K7 — wgrad with fused bias-grad: gW[K,F] = deltaᵀ[K,N] @ act[N,F],
// gB[K] = Σ_n delta[n,:].  N is huge (the 1M-flow shard) and K,F are tiny
// (≤128), so the rocBLAS heuristic (no split-K → 2 workgroups!) collapses;
// this kernel splits N across ~768 blocks, stages TRANSPOSED 32-row tiles of
// both operands in LDS (bank-padded rows), runs bf16 MFMA over the n-depth,
// and atomically accumulates fp32 partials. The bias gradient rides along as
// an extra all-ones act column (out column F), so no separate reduce pass.
// ---------------------------------------------------------------------------

// LDS row strides: +8 bf16 keeps rows 16B-aligned and de-patterns banks
#define WG_PAD 8

extern "C" __global__ __launch_bounds__(256) void flowml_wgrad_bias_bf16_v2(
    const __bf16* __restrict__ delta,  // N×K
    const __bf16* __restrict__ act,    // N×F
    float* __restrict__ out,           // K×(F+1) fp32 (zeroed; col F = gB)
    long long N, int K, int F, int Kp, int Fp,  // 16-padded; Fp covers F+1
    int waves_k, int waves_f,          // wave grid (waves_k*waves_f = 4)
    long long chunk) {                 // rows per block
  // row-major 32-row tiles (vectorized coalesced staging); fragments are
  // gathered TRANSPOSED by scalar LDS reads (no scatter on the write side)
  extern __shared__ __bf16 s_t[];  // dt[32][Kp+PAD] then at[32][Fp+PAD]
  const int dstride = Kp + WG_PAD;
  const int astride = Fp + WG_PAD;
  __bf16* dt = s_t;
  __bf16* at = s_t + 32 * dstride;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  for (int i = tid; i < 32 * (dstride + astride); i += blockDim.x)
    s_t[i] = (__bf16)0.f;
  __syncthreads();

  const int nk16 = Kp / 16;
  const int nf16 = Fp / 16;
  const int kfrag = (nk16 + waves_k - 1) / waves_k;  // <= 4 (host guarantees)
  const int ffrag = (nf16 + waves_f - 1) / waves_f;  // <= 8
  const int wk = wave / waves_f;
  const int wf = wave % waves_f;
  const int kf0 = wk * kfrag;
  const int ff0 = wf * ffrag;

  f32x4 acc[4][8];
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int b = 0; b < 8; ++b) acc[a][b] = {0.f, 0.f, 0.f, 0.f};

  const long long n_lo = blockIdx.x * chunk;
  const long long n_hi = min(n_lo + chunk, N);
  const int lrow = lane & 15;        // fragment k/f index within 16
  const int lgrp = (lane >> 4) * 8;  // fragment n offset of this lane

  const int kvec = K >> 3;           // bf16x8 chunks per delta row (K%8 tail below)
  const int ktail = K & 7;
  const int fvec = F >> 3;
  const int ftail = F & 7;

  for (long long n0 = n_lo; n0 < n_hi; n0 += 32) {
    const int nrows = (int)min((long long)32, n_hi - n0);
    __syncthreads();  // previous fragments consumed
    // coalesced row-major staging, 16 B per thread-op
    for (int i = tid; i < 32 * kvec; i += blockDim.x) {
      const int n = i / kvec;
      const int c = i % kvec;
      bf16x8 v = (n < nrows)
          ? *reinterpret_cast<const bf16x8*>(delta + (n0 + n) * (long long)K + c * 8)
          : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      *reinterpret_cast<bf16x8*>(dt + n * dstride + c * 8) = v;
    }
    if (ktail) {
      for (int i = tid; i < 32 * ktail; i += blockDim.x) {
        const int n = i / ktail;
        const int k = kvec * 8 + i % ktail;
        dt[n * dstride + k] =
            (n < nrows) ? delta[(n0 + n) * (long long)K + k] : (__bf16)0.f;
      }
    }
    for (int i = tid; i < 32 * fvec; i += blockDim.x) {
      const int n = i / fvec;
      const int c = i % fvec;
      bf16x8 v = (n < nrows)
          ? *reinterpret_cast<const bf16x8*>(act + (n0 + n) * (long long)F + c * 8)
          : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      *reinterpret_cast<bf16x8*>(at + n * astride + c * 8) = v;
    }
    if (ftail) {
      for (int i = tid; i < 32 * ftail; i += blockDim.x) {
        const int n = i / ftail;
        const int f = fvec * 8 + i % ftail;
        at[n * astride + f] =
            (n < nrows) ? act[(n0 + n) * (long long)F + f] : (__bf16)0.f;
      }
    }
    if (tid < 32) at[tid * astride + F] = (tid < nrows) ? (__bf16)1.f : (__bf16)0.f;
    __syncthreads();
    // transposed fragment gathers: lane reads its 8 n-rows at a fixed k/f col
#pragma unroll
    for (int a = 0; a < 4; ++a) {
      if (a >= kfrag || kf0 + a >= nk16) continue;
      const int kcol = (kf0 + a) * 16 + lrow;
      bf16x8 afrag;
#pragma unroll
      for (int j = 0; j < 8; ++j) afrag[j] = dt[(lgrp + j) * dstride + kcol];
#pragma unroll
      for (int b = 0; b < 8; ++b) {
        if (b >= ffrag || ff0 + b >= nf16) continue;
        const int fcol = (ff0 + b) * 16 + lrow;
        bf16x8 bfrag;
#pragma unroll
        for (int j = 0; j < 8; ++j) bfrag[j] = at[(lgrp + j) * astride + fcol];
        acc[a][b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[a][b], 0, 0, 0);
      }
    }
  }

  // epilogue: D col = lane&15 (f), row = (lane>>4)*4 + reg (k)
  const int col = lane & 15;
  const int row_off = (lane >> 4) * 4;
#pragma unroll
  for (int a = 0; a < 4; ++a) {
    if (a >= kfrag || kf0 + a >= nk16) continue;
#pragma unroll
    for (int b = 0; b < 8; ++b) {
      if (b >= ffrag || ff0 + b >= nf16) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int k = (kf0 + a) * 16 + row_off + reg;
        const int f = (ff0 + b) * 16 + col;
        if (k < K && f <= F) {
          atomicAdd(&out[(long long)k * (F + 1) + f], acc[a][b][reg]);
        }
      }
    }
  }
}

template <int KF, int FF, bool PARTIAL>
__global__ void flowml_wgrad_v3(const __bf16*, const __bf16*, float*,
                                long long, int, int, int, int, int, int,
                                long long);

// exported so the binding can size the partials buffer before launching
extern "C" void wgrad_bias_plan(long long N, int* grid_out,
                                long long* chunk_out, int* version_out) {
  static const int blocks = flowml_env_int_mlp("FLOWML_WGRAD_BLOCKS", 1536);
  static const int version = flowml_env_int_mlp("FLOWML_WGRAD_V", 4);
  long long chunk = (N + blocks - 1) / blocks;
  chunk = (chunk + 31) / 32 * 32;
  *chunk_out = chunk;
  *grid_out = (int)((N + chunk - 1) / chunk);
  *version_out = version;
}

extern "C" void launch_wgrad_bias_bf16(const void* delta, const void* act,
                                       float* out, long long N, int K, int F,
                                       int grid, long long chunk, int version,
                                       void* stream) {
  auto pad16 = [](int v) { return (v + 15) / 16 * 16; };
  const int Kp = pad16(K);
  const int Fp = pad16(F + 1);
  const int nk16 = Kp / 16;
  const int nf16 = Fp / 16;
  int waves_k = 2, waves_f = 2;
  auto ok = [&](int wk2, int wf2) {
    return (nk16 + wk2 - 1) / wk2 <= 4 && (nf16 + wf2 - 1) / wf2 <= 8;
  };
  if (!ok(2, 2)) {
    if (ok(1, 4)) {
      waves_k = 1;
      waves_f = 4;
    } else {
      waves_k = 4;
      waves_f = 1;
    }
  }
  if (version == 2) {
    const size_t lds =
        (size_t)32 * (Kp + Fp + 2 * WG_PAD) * sizeof(__bf16);
    hipLaunchKernelGGL(flowml_wgrad_bias_bf16_v2, dim3(grid), dim3(256), lds,
                       (hipStream_t)stream, (const __bf16*)delta,
                       (const __bf16*)act, out, N, K, F, Kp, Fp, waves_k,
                       waves_f, chunk);
    return;
  }
  const size_t lds = (size_t)32 * (Kp + Fp) * sizeof(__bf16);
  const int kfrag = (nk16 + waves_k - 1) / waves_k;
  const int ffrag = (nf16 + waves_f - 1) / waves_f;
#define WG3_LAUNCH(KF, FF, P)                                                  \
  hipLaunchKernelGGL((flowml_wgrad_v3<KF, FF, P>), dim3(grid), dim3(256), lds, \
                     (hipStream_t)stream, (const __bf16*)delta,                \
                     (const __bf16*)act, out, N, K, F, Kp, Fp, waves_k,        \
                     waves_f, chunk)
#define WG3_PICK(P)                                        \
  if (kfrag <= 1 && ffrag <= 8) WG3_LAUNCH(1, 8, P);       \
  else if (kfrag <= 2 && ffrag <= 6) WG3_LAUNCH(2, 6, P);  \
  else if (kfrag <= 4 && ffrag <= 4) WG3_LAUNCH(4, 4, P);  \
  else WG3_LAUNCH(4, 8, P)
  if (version >= 4) { WG3_PICK(true); } else { WG3_PICK(false); }
#undef WG3_PICK
#undef WG3_LAUNCH
}

// ---------------------------------------------------------------------------
// K7f32 — deterministic split-K f32 wgrad for tall-skinny outer products
// (LR gradient: gradW[c,f] = Σ_r D[r,c]·X[r,f], gradB[c] = Σ_r D[r,c]).
// rocBLAS picks a no-split-K tile for C×F×N = 5×42×1.25M (2 workgroups on a
// 256-CU chip); here ~768 blocks each reduce a row-chunk through an LDS tile
// and write an EXACT partial (fixed sequential order, no atomics) to
// part[s, c*(F+1)+f]; the host sums over s in fixed order → deterministic.
// Bias grad fused as implicit all-ones column f == F.
// ---------------------------------------------------------------------------

#define WGF32_TILE 128

extern "C" __global__ void flowml_wgrad_f32_splitk(
    const float* __restrict__ X,  // N×F
    const float* __restrict__ D,  // N×C
    float* __restrict__ part,     // S×(C·(F+1))
    long long N, int F, int C, long long rows_per_block) {
  extern __shared__ float lds[];
  const int Fp = F + 1;
  const int FC = C * Fp;
  const int ncell = (FC + (int)blockDim.x - 1) / (int)blockDim.x;  // ≤4 enforced by launcher
  float acc[4] = {0.f, 0.f, 0.f, 0.f};
  const long long r0 = (long long)blockIdx.x * rows_per_block;
  const long long r1 = min(r0 + rows_per_block, N);
  float* ldsX = lds;                   // TILE×F
  float* ldsD = lds + WGF32_TILE * F;  // TILE×C
  for (long long base = r0; base < r1; base += WGF32_TILE) {
    const int rows = (int)min((long long)WGF32_TILE, r1 - base);
    __syncthreads();
    for (int i = threadIdx.x; i < rows * F; i += blockDim.x)
      ldsX[i] = X[base * F + i];
    for (int i = threadIdx.x; i < rows * C; i += blockDim.x)
      ldsD[i] = D[base * C + i];
    __syncthreads();
    for (int k = 0; k < 4; ++k) {
      if (k >= ncell) break;
      const int cell = (int)threadIdx.x + k * (int)blockDim.x;
      if (cell >= FC) break;
      const int c = cell / Fp, f = cell % Fp;
      float s = acc[k];
      if (f == F) {
        for (int r = 0; r < rows; ++r) s += ldsD[r * C + c];
      } else {
        for (int r = 0; r < rows; ++r) s += ldsX[r * F + f] * ldsD[r * C + c];
      }
      acc[k] = s;
    }
  }
  float* dst = part + (long long)blockIdx.x * FC;
  for (int k = 0; k < 4; ++k) {
    if (k >= ncell) break;
    const int cell = (int)threadIdx.x + k * (int)blockDim.x;
    if (cell < FC) dst[cell] = acc[k];
  }
}

// caller (ext.cpp) picks grid/chunk so it can size `part` first:
//   chunk = max(ceil(N/768), 128); grid = ceil(N/chunk); FC = C*(F+1) ≤ 1024
extern "C" void launch_wgrad_f32_splitk(const float* X, const float* D,
                                        float* part, long long N, int F, int C,
                                        int grid, long long chunk, void* stream) {
  const size_t lds_b = (size_t)WGF32_TILE * (F + C) * sizeof(float);
  hipLaunchKernelGGL(flowml_wgrad_f32_splitk, dim3(grid), dim3(256), lds_b,
                     (hipStream_t)stream, X, D, part, N, F, C, chunk);
}

// ---------------------------------------------------------------------------
// K7 v3 — same split-N wgrad, fragments loaded with the gfx950 hardware
// transpose read. v2's transposed SCALAR gathers are the measured limiter
// (96 ds_read_u16/tile/wave with 4-way group conflicts; all three layer
// shapes flatline at ~327 µs). v3 stages each operand into per-16-col tiles
// whose 32 rows are PERMUTED so that `ds_read_b64_tr_b16` (per 16-lane
// group: a contiguous 128 B block read as a 4×16 bf16 matrix, lane i gets
// column i — mapping verified empirically, scripts/tr_probe.hip) delivers
// an MFMA fragment in 2 instructions instead of 8 scalar reads per bf16x8.
// Tile layout: elem(n, c) = tile*512 + ((n>>2)&1)*256 + (n>>3)*64 + (n&3)*16 + c
// so window w reads rows n = 8g + 4w + j for lane group g, elem j.
// ---------------------------------------------------------------------------

typedef __bf16 bf16x4_t __attribute__((ext_vector_type(4)));

__device__ __forceinline__ int wg3_elem(int n, int c) {
  return ((n >> 2) & 1) * 256 + (n >> 3) * 64 + (n & 3) * 16 + c;
}

// Templated on the per-wave fragment grid: a static acc[4][8] allocated
// 252 VGPR + 128 AGPR = 1 wave/SIMD (no latency hiding). Instantiating the
// exact (KF, FF) needed by each layer shape keeps the accumulators small
// enough for 2-3 waves/SIMD.
template <int KF, int FF, bool PARTIAL>
__global__ __launch_bounds__(256) void flowml_wgrad_v3(
    const __bf16* __restrict__ delta,  // N×K
    const __bf16* __restrict__ act,    // N×F
    float* __restrict__ out,  // PARTIAL: grid×K×(F+1) exact per-block slabs
                              // (zeroed; summed host-side in fixed order ⇒
                              // DETERMINISTIC); else K×(F+1) atomic target
    long long N, int K, int F, int Kp, int Fp,
    int waves_k, int waves_f, long long chunk) {
  extern __shared__ __bf16 s_t[];  // dt tiles [nk16][512] then at [nf16][512]
  const int nk16 = Kp / 16;
  const int nf16 = Fp / 16;
  __bf16* dt = s_t;
  __bf16* at = s_t + nk16 * 512;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  for (int i = tid; i < (nk16 + nf16) * 512; i += blockDim.x)
    s_t[i] = (__bf16)0.f;
  __syncthreads();

  const int kfrag = (nk16 + waves_k - 1) / waves_k;  // <= 4
  const int ffrag = (nf16 + waves_f - 1) / waves_f;  // <= 8
  const int wk = wave / waves_f;
  const int wf = wave % waves_f;
  const int kf0 = wk * kfrag;
  const int ff0 = wf * ffrag;

  f32x4 acc[KF][FF];
#pragma unroll
  for (int a = 0; a < KF; ++a)
#pragma unroll
    for (int b = 0; b < FF; ++b) acc[a][b] = {0.f, 0.f, 0.f, 0.f};

  const long long n_lo = blockIdx.x * chunk;
  const long long n_hi = min(n_lo + chunk, N);
  const int kvec = K >> 3;
  const int ktail = K & 7;
  const int fvec = F >> 3;
  const int ftail = F & 7;

  auto dt3 = (__attribute__((address_space(3))) bf16x4_t*)dt;
  auto at3 = (__attribute__((address_space(3))) bf16x4_t*)at;

  for (long long n0 = n_lo; n0 < n_hi; n0 += 32) {
    const int nrows = (int)min((long long)32, n_hi - n0);
    __syncthreads();
    // staging: 8-elem (16 B) pieces land inside one tile row (c0 % 8 == 0)
    for (int i = tid; i < 32 * kvec; i += blockDim.x) {
      const int n = i / kvec;
      const int c = i % kvec;
      bf16x8 v = (n < nrows)
          ? *reinterpret_cast<const bf16x8*>(delta + (n0 + n) * (long long)K + c * 8)
          : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      const int c0 = c * 8;
      *reinterpret_cast<bf16x8*>(dt + (c0 >> 4) * 512 + wg3_elem(n, c0 & 15)) = v;
    }
    if (ktail) {
      for (int i = tid; i < 32 * ktail; i += blockDim.x) {
        const int n = i / ktail;
        const int k = kvec * 8 + i % ktail;
        dt[(k >> 4) * 512 + wg3_elem(n, k & 15)] =
            (n < nrows) ? delta[(n0 + n) * (long long)K + k] : (__bf16)0.f;
      }
    }
    for (int i = tid; i < 32 * fvec; i += blockDim.x) {
      const int n = i / fvec;
      const int c = i % fvec;
      bf16x8 v = (n < nrows)
          ? *reinterpret_cast<const bf16x8*>(act + (n0 + n) * (long long)F + c * 8)
          : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      const int c0 = c * 8;
      *reinterpret_cast<bf16x8*>(at + (c0 >> 4) * 512 + wg3_elem(n, c0 & 15)) = v;
    }
    if (ftail) {
      for (int i = tid; i < 32 * ftail; i += blockDim.x) {
        const int n = i / ftail;
        const int f = fvec * 8 + i % ftail;
        at[(f >> 4) * 512 + wg3_elem(n, f & 15)] =
            (n < nrows) ? act[(n0 + n) * (long long)F + f] : (__bf16)0.f;
      }
    }
    if (tid < 32)
      at[(F >> 4) * 512 + wg3_elem(tid, F & 15)] =
          (tid < nrows) ? (__bf16)1.f : (__bf16)0.f;
    __syncthreads();

    // hoisted B fragments: 2 tr reads each, reused across the a-loop
    bf16x8 bfr[FF];
#pragma unroll
    for (int b = 0; b < FF; ++b) {
      if (b >= ffrag || ff0 + b >= nf16) continue;
      const int base = (ff0 + b) * 128;  // tile offset in bf16x4 units
      bf16x4_t u0 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(at3 + base + lane);
      bf16x4_t u1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(at3 + base + 64 + lane);
      bf16x8 f8;
#pragma unroll
      for (int j = 0; j < 4; ++j) { f8[j] = u0[j]; f8[4 + j] = u1[j]; }
      bfr[b] = f8;
    }
#pragma unroll
    for (int a = 0; a < KF; ++a) {
      if (a >= kfrag || kf0 + a >= nk16) continue;
      const int base = (kf0 + a) * 128;
      bf16x4_t v0 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(dt3 + base + lane);
      bf16x4_t v1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(dt3 + base + 64 + lane);
      bf16x8 af;
#pragma unroll
      for (int j = 0; j < 4; ++j) { af[j] = v0[j]; af[4 + j] = v1[j]; }
#pragma unroll
      for (int b = 0; b < FF; ++b) {
        if (b >= ffrag || ff0 + b >= nf16) continue;
        acc[a][b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bfr[b], acc[a][b], 0, 0, 0);
      }
    }
  }

  const int col = lane & 15;
  const int row_off = (lane >> 4) * 4;
  float* dst = PARTIAL ? out + (long long)blockIdx.x * K * (F + 1) : out;
#pragma unroll
  for (int a = 0; a < KF; ++a) {
    if (a >= kfrag || kf0 + a >= nk16) continue;
#pragma unroll
    for (int b = 0; b < FF; ++b) {
      if (b >= ffrag || ff0 + b >= nf16) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int k = (kf0 + a) * 16 + row_off + reg;
        const int f = (ff0 + b) * 16 + col;
        if (k < K && f <= F) {
          if (PARTIAL) {
            dst[(long long)k * (F + 1) + f] = acc[a][b][reg];
          } else {
            atomicAdd(&dst[(long long)k * (F + 1) + f], acc[a][b][reg]);
          }
        }
      }
    }
  }
}
