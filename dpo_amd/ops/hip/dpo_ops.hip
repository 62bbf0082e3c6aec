// dpo_amd HIP/CDNA4 kernels (gfx950 / MI355X) — fp64 core math.
//
// Design notes (MI355X-first):
//  * The per-agent pose-graph problems are small (N = (d+1)n scalar rows,
//    typically 1e3..1e5; the whole working set is L2/LLC-resident), so the
//    kernels are optimized for LATENCY and FUSION, not TFLOPs: every
//    elementwise/projection pass fuses its reductions (atomicAdd into a
//    device control block) so the truncated-CG loop needs no host round
//    trips (the C++ orchestrator enqueues guarded kernels; control
//    decisions run in single-wave kernels on-device).
//  * Layout: X is (N, r) row-major fp64 ("Xt layout"): pose i occupies
//    rows [i*dh, i*dh+dh), the first d of which are the transposed
//    Stiefel block, the last the translation. Consecutive lanes read
//    consecutive (row, col) elements -> coalesced.
//  * Q is (d+1)x(d+1) block-CSR ("BSR"): row_ptr (n+1), col_idx (nnzb),
//    vals (nnzb, dh, dh). The standalone SpMM assigns a pose-block row
//    per dh*r-thread group (one output element per thread); the FUSED
//    Hessian/projection kernel computes a whole pose-block row per
//    thread with the dh x r accumulator tile in registers. fp64 FMA;
//    bandwidth/latency-bound at these sizes.
//  * Wavefront = 64; blocks are multiples of 64 threads.
//
// Functional parity targets: Hessian-vec V*Q
// (reference QuadraticProblem.cpp:68-73), tangent projection / polar
// retraction (ROPTLIB Stiefel ops), preconditioner apply
// (QuadraticProblem.cpp:75-87), G/Q assembly (PGOAgent.cpp:720-859).

#include <hip/hip_runtime.h>
#include <math.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

// Debug/bisection knobs: DPO_NO_SOLVE_GRAPH=1 / DPO_NO_EVAL_GRAPH=1 run
// the solve / eval sequences eagerly instead of via cached hipGraphs.
// teardown/cleanup calls whose errors are deliberately ignored
static inline void dpo_ignore(hipError_t) {}

static bool dpo_env_flag(const char* name) {
  const char* v = getenv(name);
  return v && v[0] == '1';
}

#define DPO_CHECK(x)                                                     \
  do {                                                                   \
    hipError_t _e = (x);                                                 \
    if (_e != hipSuccess) {                                              \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(_e),  \
              __FILE__, __LINE__);                                       \
    }                                                                    \
  } while (0)

// ---------------------------------------------------------------------
// control block layout (fp64 slots) shared with the C++ orchestrator
// ---------------------------------------------------------------------
enum CtrlSlot {
  C_STATUS = 0,   // 0 run, 1 tcg_stop, 2 accepted, 3 give_up, 4 no_update
  C_FX = 1,
  C_GN0SQ = 2,
  C_RR = 3,
  C_ZR = 4,
  C_EPE = 5,
  C_EPD = 6,
  C_DPD = 7,
  C_COEF = 8,     // alpha or tau applied in the current eta update
  C_BETA = 9,
  C_ITER = 10,
  C_J = 11,       // snapshots stored (completed tCG iterations)
  C_RADIUS = 12,
  C_FPROP = 13,
  C_DM = 14,      // model decrease of current candidate step
  C_JSTAR = 15,
  C_TAUSTAR = 16,
  C_USE_CURRENT = 17,  // candidate step == current eta (converged tCG)
  C_STOP_PENDING = 18, // boundary/negcurv hit in current iteration
  C_BOUND = 19,        // tCG residual bound
  C_GN1SQ = 20,
  C_RHO = 21,
  C_HLEN = 22,     // valid Krylov-history entries (radius-independent)
  C_SHRINKS = 23,  // radius-shrink attempts taken (batched shrink loop)
  C_DOT0 = 24,   // scratch dot slots (cleared by ctrl kernels)
  C_DOT1 = 25,
  C_DOT2 = 26,
  C_DOT3 = 27,
  C_HIST = 32,   // 5 arrays of length MAX_TCG: z_r, d_Hd, e_Pe, e_Pd, d_Pd
};
#define MAX_TCG 16
#define H_ZR(j) (C_HIST + (j))
#define H_DHD(j) (C_HIST + MAX_TCG + (j))
#define H_EPE(j) (C_HIST + 2 * MAX_TCG + (j))
#define H_EPD(j) (C_HIST + 3 * MAX_TCG + (j))
#define H_DPD(j) (C_HIST + 4 * MAX_TCG + (j))
#define CTRL_SIZE (C_HIST + 5 * MAX_TCG)

enum Status {
  ST_RUN = 0,
  ST_TCG_STOP = 1,
  ST_ACCEPTED = 2,
  ST_GIVE_UP = 3,
  ST_NO_UPDATE = 4,
};

// guard < 0: no guard; else kernel runs only when status == guard.
__device__ __forceinline__ bool guarded_off(const double* ctrl, int guard) {
  return guard >= 0 && ctrl != nullptr && ctrl[C_STATUS] != (double)guard;
}

#define C_TICKET C_DOT3  // fan-in arrival counter (scratch slot)

// L2-coherent (L1-bypassing) ctrl accesses for the fused control tails:
// other workgroups' dot contributions arrive as device-scope atomics, so
// the last-arriving block must read them past its own L1.
__device__ __forceinline__ double ctrl_load(const double* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}
__device__ __forceinline__ void ctrl_store(double* p, double v) {
  __hip_atomic_store(p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

// Returns true for exactly one block — the last to arrive — after every
// block's prior vector-memory ops (incl. the dot atomics) have retired.
// Split-K fan-in (guide G16): per-block vmcnt drain, then a device-scope
// ticket; the winner resets the ticket for the next fused kernel.
__device__ __forceinline__ bool fanin_last_block(double* ctrl) {
  __syncthreads();
  __shared__ int is_last;
  if (threadIdx.x == 0) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    double old = atomicAdd(ctrl + C_TICKET, 1.0);
    is_last = ((int)old == (int)gridDim.x - 1) ? 1 : 0;
    if (is_last) ctrl_store(ctrl + C_TICKET, 0.0);
  }
  __syncthreads();
  return is_last != 0;
}

// Control tails (single thread of the last block). Mirror the k_ctrl_*
// kernels; dot slots are read L2-coherently, results stored L2-visibly
// (the next kernel's dispatch acquire makes them visible to all CUs).
enum CtrlFuse { CF_NONE = 0, CF_Z0 = 1, CF_ALPHA = 2, CF_RR = 3,
                CF_BETA = 4, CF_COMBINE = 5 };

__device__ void ctrl_tail_z0(double* ctrl) {
  if (ctrl[C_STATUS] != (double)ST_RUN) return;
  const double zr = ctrl_load(ctrl + C_DOT0);
  ctrl_store(ctrl + C_ZR, zr);
  ctrl_store(ctrl + C_DPD, zr);
  ctrl_store(ctrl + C_DOT0, 0.0);
}

__device__ void ctrl_tail_alpha(double* ctrl) {
  if (ctrl[C_STATUS] != (double)ST_RUN) return;
  const int j = (int)ctrl[C_ITER];
  const double d_Hd = ctrl_load(ctrl + C_DOT0);
  ctrl_store(ctrl + C_DOT0, 0.0);
  const double z_r = ctrl[C_ZR];
  const double e_Pe = ctrl[C_EPE], e_Pd = ctrl[C_EPD], d_Pd = ctrl[C_DPD];
  const double radius = ctrl[C_RADIUS];
  ctrl_store(ctrl + H_ZR(j), z_r);
  ctrl_store(ctrl + H_DHD(j), d_Hd);
  ctrl_store(ctrl + H_EPE(j), e_Pe);
  ctrl_store(ctrl + H_EPD(j), e_Pd);
  ctrl_store(ctrl + H_DPD(j), d_Pd);
  const double alpha = z_r / d_Hd;
  const double e_Pe_new = e_Pe + 2.0 * alpha * e_Pd + alpha * alpha * d_Pd;
  if (d_Hd <= 0.0 || e_Pe_new >= radius * radius) {
    const double disc = e_Pd * e_Pd + d_Pd * (radius * radius - e_Pe);
    const double tau = (d_Pd > 0.0)
        ? (-e_Pd + sqrt(fmax(disc, 0.0))) / d_Pd : 0.0;
    ctrl_store(ctrl + C_COEF, tau);
    ctrl_store(ctrl + C_STOP_PENDING, 1.0);
  } else {
    ctrl_store(ctrl + C_COEF, alpha);
    ctrl_store(ctrl + C_EPE, e_Pe_new);
    ctrl_store(ctrl + C_STOP_PENDING, 0.0);
  }
}

__device__ void ctrl_tail_rr(double* ctrl) {
  if (ctrl[C_STATUS] != (double)ST_RUN) return;
  const int j = (int)ctrl[C_ITER];
  if (ctrl[C_STOP_PENDING] != 0.0) {
    ctrl_store(ctrl + C_STATUS, (double)ST_TCG_STOP);
    ctrl_store(ctrl + C_J, (double)j);
    ctrl_store(ctrl + C_HLEN, (double)(j + 1));
    ctrl_store(ctrl + C_USE_CURRENT, 0.0);
    return;
  }
  const double rr = ctrl_load(ctrl + C_DOT1);
  ctrl_store(ctrl + C_DOT1, 0.0);
  ctrl_store(ctrl + C_RR, rr);
  if (sqrt(rr) <= ctrl[C_BOUND]) {
    ctrl_store(ctrl + C_STATUS, (double)ST_TCG_STOP);
    ctrl_store(ctrl + C_J, (double)(j + 1));
    ctrl_store(ctrl + C_HLEN, (double)(j + 1));
    ctrl_store(ctrl + C_USE_CURRENT, 1.0);
  }
}

__device__ void ctrl_tail_beta(double* ctrl) {
  if (ctrl[C_STATUS] != (double)ST_RUN) return;
  const double z_r_new = ctrl_load(ctrl + C_DOT0);
  ctrl_store(ctrl + C_DOT0, 0.0);
  const double z_r = ctrl[C_ZR];
  const double alpha = ctrl[C_COEF];
  const double beta = z_r_new / z_r;
  ctrl_store(ctrl + C_BETA, beta);
  ctrl_store(ctrl + C_EPD, beta * (ctrl[C_EPD] + alpha * ctrl[C_DPD]));
  ctrl_store(ctrl + C_DPD, z_r_new + beta * beta * ctrl[C_DPD]);
  ctrl_store(ctrl + C_ZR, z_r_new);
  ctrl_store(ctrl + C_ITER, ctrl[C_ITER] + 1.0);
}

__device__ void ctrl_tail_combine(double* ctrl, double* out) {
  out[0] = 0.5 * (ctrl_load(ctrl + C_DOT0) + ctrl_load(ctrl + C_DOT2));
  out[1] = 0.5 * ctrl_load(ctrl + C_DOT2);
  out[2] = ctrl_load(ctrl + C_DOT1);
}

__device__ __forceinline__ void run_ctrl_tail(int cf, double* ctrl,
                                              double* out) {
  switch (cf) {
    case CF_Z0: ctrl_tail_z0(ctrl); break;
    case CF_ALPHA: ctrl_tail_alpha(ctrl); break;
    case CF_RR: ctrl_tail_rr(ctrl); break;
    case CF_BETA: ctrl_tail_beta(ctrl); break;
    case CF_COMBINE: ctrl_tail_combine(ctrl, out); break;
    default: break;
  }
}

// wave-level + block-level reduction, then one atomicAdd per block
__device__ __forceinline__ void block_reduce_atomic(double v, double* dst) {
  __shared__ double sh[16];  // up to 1024/64 waves
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_down(v, off, 64);
  if (lane == 0) sh[wave] = v;
  __syncthreads();
  int nwaves = (blockDim.x + 63) >> 6;
  if (wave == 0) {
    double s = (lane < nwaves) ? sh[lane] : 0.0;
    for (int off = 32; off > 0; off >>= 1)
      s += __shfl_down(s, off, 64);
    if (lane == 0 && s != 0.0) atomicAdd(dst, s);
  }
  // the scratch is reused by back-to-back reductions in one kernel:
  // wave 0 must finish reading before anyone writes the next round
  __syncthreads();
}

// ---------------------------------------------------------------------
// BSR SpMM: out(N, r) = Q(N, N) @ X(N, r)
// One dh*r-thread group per pose-block row; each thread owns one
// (row_in_block, col) output element and accumulates over the row's
// blocks. Q block values are broadcast from L2 (tiny per-agent Q is
// cache-resident); X block reads are contiguous dh*r segments.
// ---------------------------------------------------------------------
template <int DH, int R>
__global__ void k_bsr_spmm(const int* __restrict__ row_ptr,
                           const int* __restrict__ col_idx,
                           const double* __restrict__ vals,
                           const double* __restrict__ X,
                           double* __restrict__ out,
                           int n, const double* __restrict__ ctrl,
                           int guard) {
  if (guarded_off(ctrl, guard)) return;
  constexpr int tile = DH * R;
  constexpr int per_block = 256 / tile;
  const int slot = threadIdx.x / tile;
  const int t = threadIdx.x % tile;
  const int i = blockIdx.x * per_block + slot;
  if (threadIdx.x >= per_block * tile || i >= n) return;
  const int c = t / R;   // row inside block
  const int k = t % R;   // column of X
  const int s = row_ptr[i], e = row_ptr[i + 1];
  double acc = 0.0;
  for (int p = s; p < e; ++p) {
    const int j = col_idx[p];
    const double* B = vals + (size_t)p * DH * DH;
    const double* Xj = X + (size_t)j * DH * R;
    #pragma unroll
    for (int cc = 0; cc < DH; ++cc)
      acc = fma(B[c * DH + cc], Xj[cc * R + k], acc);
  }
  out[(size_t)i * DH * R + c * R + k] = acc;
}

// generic-(dh, r) fallback (rare shapes); compile-time tile kernels above
__global__ void k_bsr_spmm_gen(const int* __restrict__ row_ptr,
                               const int* __restrict__ col_idx,
                               const double* __restrict__ vals,
                               const double* __restrict__ X,
                               double* __restrict__ out,
                               int n, int dh, int r,
                               const double* __restrict__ ctrl, int guard) {
  if (guarded_off(ctrl, guard)) return;
  const int tile = dh * r;
  const int per_block = blockDim.x / tile;
  const int slot = threadIdx.x / tile;
  const int t = threadIdx.x % tile;
  const int i = blockIdx.x * per_block + slot;
  if (threadIdx.x >= per_block * tile || i >= n) return;
  const int c = t / r;
  const int k = t % r;
  const int s = row_ptr[i], e = row_ptr[i + 1];
  double acc = 0.0;
  for (int p = s; p < e; ++p) {
    const int j = col_idx[p];
    const double* B = vals + (size_t)p * dh * dh;
    const double* Xj = X + (size_t)j * dh * r;
    for (int cc = 0; cc < dh; ++cc)
      acc = fma(B[c * dh + cc], Xj[cc * r + k], acc);
  }
  out[(size_t)i * dh * r + c * r + k] = acc;
}

// ---------------------------------------------------------------------
// Fused tangent projection + dots.
//   P = V (+ G) projected onto T_X; optionally negate; write to out.
//   Optional atomic dots: <P, dotWith> -> ctrl[dot_slot]
//                         <V+G, X> -> ctrl[dot_slot2]   (for f(X))
// One thread per pose; all of Yt (d x r), Vt (dh x r) live in registers
// (r <= DPO_MAX_R, dh <= 4).
// ---------------------------------------------------------------------
#define DPO_MAX_R 8
#define DPO_MAX_DH 4

template <int NEG, int D, int R, int CF = CF_NONE>
__global__ void k_proj_dots(const double* __restrict__ X,
                            const double* __restrict__ V,
                            const double* __restrict__ G,
                            double* __restrict__ out,
                            const double* __restrict__ dotWith,
                            double* __restrict__ ctrl,
                            int n, int dot_slot, int dot_slot2,
                            int guard) {
  if (guarded_off(ctrl, guard)) return;
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  constexpr int dh = D + 1;
  double Yt[D][R];
  double Vt[dh][R];
  double dot_vx = 0.0;
  if (i < n) {
    const double* Xi = X + (size_t)i * dh * R;
    const double* Vi = V + (size_t)i * dh * R;
    const double* Gi = G ? G + (size_t)i * dh * R : nullptr;
    #pragma unroll
    for (int c = 0; c < dh; ++c)
      #pragma unroll
      for (int k = 0; k < R; ++k) {
        double v = Vi[c * R + k];
        if (Gi) v += Gi[c * R + k];
        Vt[c][k] = v;
        if (dot_slot2 >= 0) dot_vx = fma(v, Xi[c * R + k], dot_vx);
      }
    #pragma unroll
    for (int c = 0; c < D; ++c)
      #pragma unroll
      for (int k = 0; k < R; ++k)
        Yt[c][k] = Xi[c * R + k];
    // S = sym(Yt Vt^T)  (D x D)
    double S[D][D];
    #pragma unroll
    for (int a = 0; a < D; ++a)
      #pragma unroll
      for (int b = 0; b < D; ++b) {
        double s = 0.0;
        #pragma unroll
        for (int k = 0; k < R; ++k) s = fma(Yt[a][k], Vt[b][k], s);
        S[a][b] = s;
      }
    #pragma unroll
    for (int a = 0; a < D; ++a)
      #pragma unroll
      for (int b = a; b < D; ++b) {
        double s = 0.5 * (S[a][b] + S[b][a]);
        S[a][b] = s;
        S[b][a] = s;
      }
    // P = Vt - S Yt on Stiefel rows
    #pragma unroll
    for (int a = 0; a < D; ++a)
      #pragma unroll
      for (int k = 0; k < R; ++k) {
        double acc = Vt[a][k];
        #pragma unroll
        for (int b = 0; b < D; ++b) acc = fma(-S[a][b], Yt[b][k], acc);
        Vt[a][k] = acc;
      }
  }
  double dot_pw = 0.0;
  if (i < n) {
    double* Oi = out + (size_t)i * dh * R;
    const double* Wi = dotWith ? dotWith + (size_t)i * dh * R : nullptr;
    #pragma unroll
    for (int c = 0; c < dh; ++c)
      #pragma unroll
      for (int k = 0; k < R; ++k) {
        double p = NEG ? -Vt[c][k] : Vt[c][k];
        Oi[c * R + k] = p;
        if (dot_slot >= 0) {
          double w = Wi ? Wi[c * R + k] : Vt[c][k];  // default: <P,P>
          dot_pw = fma(p, w, dot_pw);
        }
      }
  }
  if (dot_slot >= 0) block_reduce_atomic(dot_pw, ctrl + dot_slot);
  if (dot_slot2 >= 0) block_reduce_atomic(dot_vx, ctrl + dot_slot2);
  if (CF != CF_NONE) {
    if (fanin_last_block(ctrl) && threadIdx.x == 0)
      run_ctrl_tail(CF, ctrl, nullptr);
  }
}

// Fused block-Jacobi preconditioner apply + tangent projection (+ dot
// + control tail): one kernel instead of the {precond, proj} pair —
// one of five serial stages in every tCG iteration, so at large N the
// removed launch+drain latency is pure win. Per pose: y = M^-1 r via
// the cached per-pose Cholesky factors, P = P_X(y), dot <P, r>.
template <int D, int R, int CF = CF_NONE>
__global__ void k_precond_proj(const double* __restrict__ L,
                               const double* __restrict__ rvec,
                               const double* __restrict__ X,
                               double* __restrict__ Z,
                               double* __restrict__ ctrl,
                               int n, int dot_slot, int guard) {
  if (guarded_off(ctrl, guard)) return;
  constexpr int dh = D + 1;
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  double Vt[dh][R];
  double dot = 0.0;
  if (i < n) {
    const double* Li = L + (size_t)i * dh * dh;
    const double* Ri = rvec + (size_t)i * dh * R;
    // per-column forward/backward triangular solves (L L^T y = r)
    #pragma unroll
    for (int k = 0; k < R; ++k) {
      double y[dh];
      #pragma unroll
      for (int a = 0; a < dh; ++a) {
        double sv = Ri[a * R + k];
        #pragma unroll
        for (int b = 0; b < dh; ++b)
          if (b < a) sv = fma(-Li[a * dh + b], y[b], sv);
        y[a] = sv / Li[a * dh + a];
      }
      #pragma unroll
      for (int a = dh - 1; a >= 0; --a) {
        double sv = y[a];
        #pragma unroll
        for (int b = 0; b < dh; ++b)
          if (b > a) sv = fma(-Li[b * dh + a], y[b], sv);
        y[a] = sv / Li[a * dh + a];
      }
      #pragma unroll
      for (int a = 0; a < dh; ++a) Vt[a][k] = y[a];
    }
    // tangent projection at X (Stiefel rows only)
    const double* Xi = X + (size_t)i * dh * R;
    double S[D][D];
    #pragma unroll
    for (int a = 0; a < D; ++a)
      #pragma unroll
      for (int b = 0; b < D; ++b) {
        double sv = 0.0;
        #pragma unroll
        for (int k = 0; k < R; ++k)
          sv = fma(Xi[a * R + k], Vt[b][k], sv);
        S[a][b] = sv;
      }
    #pragma unroll
    for (int a = 0; a < D; ++a)
      #pragma unroll
      for (int b = a; b < D; ++b) {
        const double sv = 0.5 * (S[a][b] + S[b][a]);
        S[a][b] = sv;
        S[b][a] = sv;
      }
    #pragma unroll
    for (int a = 0; a < D; ++a)
      #pragma unroll
      for (int k = 0; k < R; ++k) {
        double v = Vt[a][k];
        #pragma unroll
        for (int b = 0; b < D; ++b) v = fma(-S[a][b], Xi[b * R + k], v);
        Vt[a][k] = v;
      }
    double* Zi = Z + (size_t)i * dh * R;
    #pragma unroll
    for (int c = 0; c < dh; ++c)
      #pragma unroll
      for (int k = 0; k < R; ++k) {
        Zi[c * R + k] = Vt[c][k];
        if (dot_slot >= 0)
          dot = fma(Vt[c][k], Ri[c * R + k], dot);
      }
  }
  if (dot_slot >= 0) block_reduce_atomic(dot, ctrl + dot_slot);
  if (CF != CF_NONE) {
    if (fanin_last_block(ctrl) && threadIdx.x == 0)
      run_ctrl_tail(CF, ctrl, nullptr);
  }
}

// Wide (element-per-thread) tangent projection for LARGE agents: same
// wave-starvation fix as k_hess_wide (a thread-per-pose kernel gets
// only n/64 waves). dh*r threads per pose; the cross-element sym(Y V^T)
// contraction goes through LDS tiles.
template <int NEG, int D, int R, int CF = CF_NONE>
__global__ void k_proj_wide(const double* __restrict__ X,
                            const double* __restrict__ V,
                            const double* __restrict__ G,
                            double* __restrict__ out,
                            const double* __restrict__ dotWith,
                            double* __restrict__ ctrl,
                            int n, int dot_slot, int dot_slot2,
                            int guard) {
  if (guarded_off(ctrl, guard)) return;
  constexpr int dh = D + 1;
  constexpr int TILE = dh * R;
  constexpr int PB = 256 / TILE;
  __shared__ double sV[PB][TILE];
  __shared__ double sX[PB][TILE];
  const int slot = threadIdx.x / TILE;
  const int e = threadIdx.x % TILE;
  const int c = e / R;
  const int k = e % R;
  const bool active = threadIdx.x < PB * TILE;
  const int i = blockIdx.x * PB + slot;
  const bool live = active && i < n;
  double v = 0.0, x = 0.0, d0 = 0.0, d1 = 0.0;
  if (live) {
    v = V[(size_t)i * TILE + e];
    if (G) v += G[(size_t)i * TILE + e];
    x = X[(size_t)i * TILE + e];
    if (dot_slot2 >= 0) d1 = v * x;
    sV[slot][e] = v;
    sX[slot][e] = x;
  }
  __syncthreads();
  double o = v;
  if (live && c < D) {
    #pragma unroll
    for (int b = 0; b < D; ++b) {
      double s1 = 0.0, s2 = 0.0;
      #pragma unroll
      for (int kk = 0; kk < R; ++kk) {
        s1 = fma(sX[slot][c * R + kk], sV[slot][b * R + kk], s1);
        s2 = fma(sX[slot][b * R + kk], sV[slot][c * R + kk], s2);
      }
      o = fma(-0.5 * (s1 + s2), sX[slot][b * R + k], o);
    }
  }
  if (live) {
    const double p = NEG ? -o : o;
    out[(size_t)i * TILE + e] = p;
    if (dot_slot >= 0) {
      const double w = dotWith ? dotWith[(size_t)i * TILE + e] : o;
      d0 = p * w;
    }
  }
  __syncthreads();
  if (dot_slot >= 0) block_reduce_atomic(d0, ctrl + dot_slot);
  if (dot_slot2 >= 0) block_reduce_atomic(d1, ctrl + dot_slot2);
  if (CF != CF_NONE) {
    if (fanin_last_block(ctrl) && threadIdx.x == 0)
      run_ctrl_tail(CF, ctrl, nullptr);
  }
}

// ---------------------------------------------------------------------
// Fused Hessian-vector product: one THREAD per pose computes its whole
// BSR row of Q @ V (the dh x r accumulator tile lives in registers; each
// Q block is one 128-byte line), optionally adds G, applies the tangent
// projection at X, writes the result and accumulates the requested dot
// products — replacing a {SpMM kernel, projection kernel} pair with one
// launch. MODE 0: out = P_X(QV + G), dots <out, dotW> / <QV+G, X>.
// MODE 1 (no projection, no store): dots <QV, V> and <G, V> (the
// trust-region candidate's f evaluation).
// ---------------------------------------------------------------------
template <int D, int R, int MODE, int CF = CF_NONE>
__global__ void k_hess_fused(const int* __restrict__ row_ptr,
                             const int* __restrict__ col_idx,
                             const double* __restrict__ vals,
                             const double* __restrict__ V,
                             const double* __restrict__ X,
                             const double* __restrict__ G,
                             double* __restrict__ out,
                             const double* __restrict__ dotW,
                             double* __restrict__ ctrl,
                             int n, int dot_slot, int dot_slot2,
                             int dot_slot3, int guard) {
  if (guarded_off(ctrl, guard)) return;
  constexpr int dh = D + 1;
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  double acc[dh][R];
  double d0 = 0.0, d1 = 0.0, d2 = 0.0;
  if (i < n) {
    #pragma unroll
    for (int c = 0; c < dh; ++c)
      #pragma unroll
      for (int k = 0; k < R; ++k) acc[c][k] = 0.0;
    const int s0 = row_ptr[i], e0 = row_ptr[i + 1];
    for (int p = s0; p < e0; ++p) {
      const double* B = vals + (size_t)p * dh * dh;
      const double* Vj = V + (size_t)col_idx[p] * dh * R;
      #pragma unroll
      for (int c = 0; c < dh; ++c)
        #pragma unroll
        for (int cc = 0; cc < dh; ++cc) {
          const double b = B[c * dh + cc];
          #pragma unroll
          for (int k = 0; k < R; ++k)
            acc[c][k] = fma(b, Vj[cc * R + k], acc[c][k]);
        }
    }
    if (G) {
      const double* Gi = G + (size_t)i * dh * R;
      const double* Xi3 = (MODE == 0 && dot_slot3 >= 0)
          ? X + (size_t)i * dh * R : nullptr;
      #pragma unroll
      for (int c = 0; c < dh; ++c)
        #pragma unroll
        for (int k = 0; k < R; ++k) {
          const double g = Gi[c * R + k];
          acc[c][k] += g;
          if (Xi3) d2 = fma(g, Xi3[c * R + k], d2);
        }
    }
    if (MODE == 1) {
      // dots only: d0 = <(QV)_i + G_i, V_i>, d1 = <G_i, V_i>
      const double* Vi = V + (size_t)i * dh * R;
      const double* Gi = G ? G + (size_t)i * dh * R : nullptr;
      #pragma unroll
      for (int c = 0; c < dh; ++c)
        #pragma unroll
        for (int k = 0; k < R; ++k) {
          const double g = Gi ? Gi[c * R + k] : 0.0;
          d0 = fma(acc[c][k] - g, Vi[c * R + k], d0);
          d1 = fma(g, Vi[c * R + k], d1);
        }
    } else {
      const double* Xi = X + (size_t)i * dh * R;
      if (dot_slot2 >= 0) {
        #pragma unroll
        for (int c = 0; c < dh; ++c)
          #pragma unroll
          for (int k = 0; k < R; ++k)
            d1 = fma(acc[c][k], Xi[c * R + k], d1);
      }
      // tangent projection at X: S = sym(Yt A^T); A -= S Yt
      double S[D][D];
      #pragma unroll
      for (int a = 0; a < D; ++a)
        #pragma unroll
        for (int b = 0; b < D; ++b) {
          double sv = 0.0;
          #pragma unroll
          for (int k = 0; k < R; ++k)
            sv = fma(Xi[a * R + k], acc[b][k], sv);
          S[a][b] = sv;
        }
      #pragma unroll
      for (int a = 0; a < D; ++a)
        #pragma unroll
        for (int b = a; b < D; ++b) {
          const double sv = 0.5 * (S[a][b] + S[b][a]);
          S[a][b] = sv;
          S[b][a] = sv;
        }
      #pragma unroll
      for (int a = 0; a < D; ++a)
        #pragma unroll
        for (int k = 0; k < R; ++k) {
          double v = acc[a][k];
          #pragma unroll
          for (int b = 0; b < D; ++b) v = fma(-S[a][b], Xi[b * R + k], v);
          acc[a][k] = v;
        }
      double* Oi = out + (size_t)i * dh * R;
      const double* Wi = dotW ? dotW + (size_t)i * dh * R : nullptr;
      #pragma unroll
      for (int c = 0; c < dh; ++c)
        #pragma unroll
        for (int k = 0; k < R; ++k) {
          Oi[c * R + k] = acc[c][k];
          if (dot_slot >= 0) {
            const double w = Wi ? Wi[c * R + k] : acc[c][k];
            d0 = fma(acc[c][k], w, d0);
          }
        }
    }
  }
  if (dot_slot >= 0) block_reduce_atomic(d0, ctrl + dot_slot);
  if (dot_slot2 >= 0) block_reduce_atomic(d1, ctrl + dot_slot2);
  if (MODE == 0 && dot_slot3 >= 0)
    block_reduce_atomic(d2, ctrl + dot_slot3);
  if (CF != CF_NONE) {
    if (fanin_last_block(ctrl) && threadIdx.x == 0)
      run_ctrl_tail(CF, ctrl, nullptr);
  }
}

// ---------------------------------------------------------------------
// Wide (element-per-thread) variant of k_hess_fused for LARGE agents.
// One thread per pose-tile ELEMENT (dh*r threads per pose, 256/tile
// poses per block): a 125k-pose agent then launches ~39k waves instead
// of the ~2k a thread-per-pose kernel gets (which is < 2 waves/SIMD on
// 256 CUs — too few to hide HBM latency, the round-1 1M-pose
// bottleneck). The SpMM phase reads each neighbor block cooperatively;
// the cross-element tangent projection stages the acc tile and X tile
// through LDS. Same math and dot-product contract as k_hess_fused.
// ---------------------------------------------------------------------
template <int D, int R, int MODE, int CF = CF_NONE>
__global__ void k_hess_wide(const int* __restrict__ row_ptr,
                            const int* __restrict__ col_idx,
                            const double* __restrict__ vals,
                            const double* __restrict__ V,
                            const double* __restrict__ X,
                            const double* __restrict__ G,
                            double* __restrict__ out,
                            const double* __restrict__ dotW,
                            double* __restrict__ ctrl,
                            int n, int dot_slot, int dot_slot2,
                            int dot_slot3, int guard) {
  if (guarded_off(ctrl, guard)) return;
  constexpr int dh = D + 1;
  constexpr int TILE = dh * R;
  constexpr int PB = 256 / TILE;  // poses per block
  __shared__ double sAcc[PB][TILE];
  __shared__ double sX[PB][TILE];
  const int slot = threadIdx.x / TILE;
  const int e = threadIdx.x % TILE;
  const int c = e / R;            // row inside the dh x R tile
  const int k = e % R;            // column
  const bool active = threadIdx.x < PB * TILE;
  const int i = blockIdx.x * PB + slot;
  const bool live = active && i < n;
  double acc = 0.0;
  double d0 = 0.0, d1 = 0.0, d2 = 0.0;
  if (live) {
    const int s0 = row_ptr[i], e0 = row_ptr[i + 1];
    for (int p = s0; p < e0; ++p) {
      const int j = col_idx[p];            // uniform across the tile
      const double* B = vals + (size_t)p * dh * dh + c * dh;
      const double* Vj = V + (size_t)j * dh * R + k;
      #pragma unroll
      for (int cc = 0; cc < dh; ++cc)
        acc = fma(B[cc], Vj[cc * R], acc);
    }
    const double g = G ? G[(size_t)i * TILE + e] : 0.0;
    acc += g;
    if (MODE == 0 && dot_slot3 >= 0 && G)
      d2 = g * X[(size_t)i * TILE + e];
    if (MODE == 1) {
      const double v = V[(size_t)i * TILE + e];
      d0 = (acc - g) * v;
      d1 = g * v;
    } else {
      const double x = X[(size_t)i * TILE + e];
      if (dot_slot2 >= 0) d1 = acc * x;
      sAcc[slot][e] = acc;
      sX[slot][e] = x;
    }
  }
  if (MODE == 0) {
    __syncthreads();
    double o = acc;
    if (live && c < D) {
      #pragma unroll
      for (int b = 0; b < D; ++b) {
        double s1 = 0.0, s2 = 0.0;
        #pragma unroll
        for (int kk = 0; kk < R; ++kk) {
          s1 = fma(sX[slot][c * R + kk], sAcc[slot][b * R + kk], s1);
          s2 = fma(sX[slot][b * R + kk], sAcc[slot][c * R + kk], s2);
        }
        o = fma(-0.5 * (s1 + s2), sX[slot][b * R + k], o);
      }
    }
    if (live) {
      out[(size_t)i * TILE + e] = o;
      if (dot_slot >= 0) {
        const double w = dotW ? dotW[(size_t)i * TILE + e] : o;
        d0 = o * w;
      }
    }
    __syncthreads();  // sAcc/sX reuse barrier for CF tail safety
  }
  if (dot_slot >= 0) block_reduce_atomic(d0, ctrl + dot_slot);
  if (dot_slot2 >= 0) block_reduce_atomic(d1, ctrl + dot_slot2);
  if (MODE == 0 && dot_slot3 >= 0)
    block_reduce_atomic(d2, ctrl + dot_slot3);
  if (CF != CF_NONE) {
    if (fanin_last_block(ctrl) && threadIdx.x == 0)
      run_ctrl_tail(CF, ctrl, nullptr);
  }
}

// ---------------------------------------------------------------------
// fp64-MFMA BSR SpMM (A/B study, round-1 VERDICT item 4; d = 3 only).
// One wave computes 4 pose rows as a 16x16 v_mfma_f64_16x16x4_f64 tile
// accumulating over the 4-row group's COLUMN UNION (grouped-ELL built
// host-side): A = the 4 poses' stacked 4x4 Q blocks for union column u
// (zeros where a pose lacks that column), B = the neighbor pose block
// V_j (cols 5..15 zero-padded). Useful-FLOP ratio is r/16 x the union
// fill (~20-30% at r=5), and gfx950's fp64 matrix rate equals the
// vector rate, so this is expected to LOSE to the scalar-FMA kernels —
// the point is to measure that, with rocprof counters, instead of
// arguing it.
// ---------------------------------------------------------------------
typedef double dpo_d4 __attribute__((ext_vector_type(4)));

__global__ void k_bsr_spmm_mfma_d3(const int* __restrict__ grp_ptr,
                                   const int* __restrict__ grp_cols,
                                   const int* __restrict__ grp_blk,
                                   const double* __restrict__ vals,
                                   const double* __restrict__ X,
                                   double* __restrict__ out,
                                   int ngroups, int n, int r) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int g = blockIdx.x * (blockDim.x >> 6) + wave;
  if (g >= ngroups) return;
  const int row16 = lane & 15;        // A row: pose-in-group*4 + brow
  const int kfrag = lane >> 4;        // K index 0..3
  const int pose_in_g = row16 >> 2;
  const int brow = row16 & 3;
  dpo_d4 acc = {0.0, 0.0, 0.0, 0.0};
  const int s0 = grp_ptr[g], e0 = grp_ptr[g + 1];
  for (int u = s0; u < e0; ++u) {
    const int j = grp_cols[u];
    const int blk = grp_blk[(size_t)u * 4 + pose_in_g];
    const double a = (blk >= 0)
        ? vals[(size_t)blk * 16 + brow * 4 + kfrag] : 0.0;
    const int col = lane & 15;
    const double b = (col < r)
        ? X[(size_t)j * 4 * r + kfrag * r + col] : 0.0;
    acc = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc, 0, 0, 0);
  }
  // D layout (probed on gfx950, scripts/mfma_probe.py): lane l, reg q
  // -> D[row = 4*q + (l>>4)][col = l&15]
  const int col = lane & 15;
  if (col < r) {
    #pragma unroll
    for (int q = 0; q < 4; ++q) {
      const int row = 4 * q + (lane >> 4);
      const int pose = g * 4 + (row >> 2);
      if (pose < n)
        out[(size_t)pose * 4 * r + (row & 3) * r + col] = acc[q];
    }
  }
}

// ---------------------------------------------------------------------
// Batched polar projection onto (St(d, r) x R^r)^n of an affine
// combination  M = ca*A + cb*B + cc*C  (B, C optional).
// polar(Mt) for the wide d x r Stiefel block via the analytic
// eigendecomposition of the d x d Gram matrix (d <= 3).
// Translations pass through the affine combination unchanged.
// One thread per pose, everything in registers.
// ---------------------------------------------------------------------
// Inverse square root of a d x d SPD Gram matrix via the coupled
// Newton-Schulz iteration (trace-scaled):  Y -> A^{1/2}, Z -> A^{-1/2}.
// Branch-free and exact in the fully-degenerate case (Gram ~ c*I), which
// is the common case for retractions (X + small step is near-Stiefel) —
// an analytic eigenvector decomposition is ill-conditioned exactly there.
template <int D>
__device__ __forceinline__ void spd_inv_sqrt(const double S[D][D],
                                             double out[D][D]) {
  double tr = 0.0;
  #pragma unroll
  for (int i = 0; i < D; ++i) tr += S[i][i];
  if (tr <= 1e-300) {
    #pragma unroll
    for (int i = 0; i < D; ++i)
      #pragma unroll
      for (int j = 0; j < D; ++j) out[i][j] = 0.0;
    return;
  }
  const double inv_s = 1.0 / tr;
  double Y[D][D], Z[D][D];
  #pragma unroll
  for (int i = 0; i < D; ++i)
    #pragma unroll
    for (int j = 0; j < D; ++j) {
      Y[i][j] = S[i][j] * inv_s;
      Z[i][j] = (i == j) ? 1.0 : 0.0;
    }
  for (int it = 0; it < 40; ++it) {
    // T = (3 I - Z Y) / 2
    double T[D][D];
    double delta = 0.0;
    #pragma unroll
    for (int i = 0; i < D; ++i)
      #pragma unroll
      for (int j = 0; j < D; ++j) {
        double acc = 0.0;
        #pragma unroll
        for (int k = 0; k < D; ++k) acc = fma(Z[i][k], Y[k][j], acc);
        T[i][j] = 0.5 * (((i == j) ? 3.0 : 0.0) - acc);
        const double dij = T[i][j] - ((i == j) ? 1.0 : 0.0);
        delta += dij * dij;
      }
    double Yn[D][D], Zn[D][D];
    #pragma unroll
    for (int i = 0; i < D; ++i)
      #pragma unroll
      for (int j = 0; j < D; ++j) {
        double ay = 0.0, az = 0.0;
        #pragma unroll
        for (int k = 0; k < D; ++k) {
          ay = fma(Y[i][k], T[k][j], ay);
          az = fma(T[i][k], Z[k][j], az);
        }
        Yn[i][j] = ay;
        Zn[i][j] = az;
      }
    #pragma unroll
    for (int i = 0; i < D; ++i)
      #pragma unroll
      for (int j = 0; j < D; ++j) {
        Y[i][j] = Yn[i][j];
        Z[i][j] = Zn[i][j];
      }
    if (delta < 1e-30) break;  // converged (T ~ I)
  }
  const double c = rsqrt(tr);
  #pragma unroll
  for (int i = 0; i < D; ++i)
    #pragma unroll
    for (int j = 0; j < D; ++j) out[i][j] = Z[i][j] * c;
}

template <int D, int R>
__global__ void k_polar_affine(const double* __restrict__ A,
                               const double* __restrict__ B,
                               const double* __restrict__ C,
                               double ca, double cb, double cc,
                               double* __restrict__ out,
                               int n, const double* __restrict__ ctrl,
                               int guard) {
  if (guarded_off(ctrl, guard)) return;
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  constexpr int dh = D + 1;
  double Mt[dh][R];
  const double* Ai = A + (size_t)i * dh * R;
  const double* Bi = B ? B + (size_t)i * dh * R : nullptr;
  const double* Ci = C ? C + (size_t)i * dh * R : nullptr;
  #pragma unroll
  for (int c = 0; c < dh; ++c)
    #pragma unroll
    for (int k = 0; k < R; ++k) {
      double v = ca * Ai[c * R + k];
      if (Bi) v = fma(cb, Bi[c * R + k], v);
      if (Ci) v = fma(cc, Ci[c * R + k], v);
      Mt[c][k] = v;
    }
  // Gram = Mt Mt^T (D x D); polar(Mt) = Gram^{-1/2} Mt
  double S[D][D];
  #pragma unroll
  for (int a = 0; a < D; ++a)
    #pragma unroll
    for (int b = 0; b < D; ++b) {
      double s = 0.0;
      #pragma unroll
      for (int k = 0; k < R; ++k) s = fma(Mt[a][k], Mt[b][k], s);
      S[a][b] = s;
    }
  double Gi[D][D];
  spd_inv_sqrt<D>(S, Gi);
  double* Oi = out + (size_t)i * dh * R;
  #pragma unroll
  for (int a = 0; a < D; ++a)
    #pragma unroll
    for (int k = 0; k < R; ++k) {
      double s = 0.0;
      #pragma unroll
      for (int b = 0; b < D; ++b) s = fma(Gi[a][b], Mt[b][k], s);
      Oi[a * R + k] = s;
    }
  #pragma unroll
  for (int k = 0; k < R; ++k) Oi[D * R + k] = Mt[D][k];
}

// ---------------------------------------------------------------------
// Dense fp32 preconditioner apply: Z(N, r) = Minv(N, N) @ V(N, r).
// Minv is symmetric -> read column-major (Minv[j*N+i]) so consecutive
// threads (i) are coalesced; j-loop broadcasts V[j][*] via LDS.
// fp32 storage halves the bandwidth; accumulate fp64.
// ---------------------------------------------------------------------
// Z = Minv @ V with Minv symmetric fp32 read column-wise (coalesced).
// The j loop is SPLIT across gridDim.y so small problems still fill the
// chip (MI355X: 256 CUs want >> 256 workgroups); partial sums combine
// with fp64 atomics into the zeroed output.
template <int R>
__global__ void k_precond_dense(const float* __restrict__ Minv,
                                const double* __restrict__ V,
                                double* __restrict__ Z,
                                int N, const double* __restrict__ ctrl,
                                int guard) {
  if (guarded_off(ctrl, guard)) return;
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  const int jsplit = gridDim.y;
  const int jchunk = (N + jsplit - 1) / jsplit;
  const int j_lo = blockIdx.y * jchunk;
  const int j_hi = min(N, j_lo + jchunk);
  constexpr int TILE_J = 64;
  __shared__ double shv[TILE_J * R];
  double acc[R];
  #pragma unroll
  for (int k = 0; k < R; ++k) acc[k] = 0.0;
  for (int j0 = j_lo; j0 < j_hi; j0 += TILE_J) {
    const int jn = min(TILE_J, j_hi - j0);
    __syncthreads();
    for (int t = threadIdx.x; t < jn * R; t += blockDim.x)
      shv[t] = V[(size_t)(j0 + t / R) * R + (t % R)];
    __syncthreads();
    if (i < N) {
      for (int j = 0; j < jn; ++j) {
        const double m = (double)Minv[(size_t)(j0 + j) * N + i];
        #pragma unroll
        for (int k = 0; k < R; ++k)
          acc[k] = fma(m, shv[j * R + k], acc[k]);
      }
    }
  }
  if (i < N) {
    if (jsplit == 1) {
      #pragma unroll
      for (int k = 0; k < R; ++k) Z[(size_t)i * R + k] = acc[k];
    } else {
      #pragma unroll
      for (int k = 0; k < R; ++k)
        atomicAdd(&Z[(size_t)i * R + k], acc[k]);
    }
  }
}

// Block-Jacobi apply: per pose solve (L L^T) z = v with stored Cholesky
// factors L (n, dh, dh). One thread per pose per rhs column.
template <int DH>
__global__ void k_precond_jacobi(const double* __restrict__ L,
                                 const double* __restrict__ V,
                                 double* __restrict__ Z,
                                 int n, int r,
                                 const double* __restrict__ ctrl, int guard) {
  if (guarded_off(ctrl, guard)) return;
  const int idx = blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= n * r) return;
  const int i = idx / r, k = idx % r;
  const double* Li = L + (size_t)i * DH * DH;
  double y[DH];
  // forward solve L y = v
  #pragma unroll
  for (int a = 0; a < DH; ++a) {
    double s = V[((size_t)i * DH + a) * r + k];
    #pragma unroll
    for (int b = 0; b < DH; ++b)
      if (b < a) s = fma(-Li[a * DH + b], y[b], s);
    y[a] = s / Li[a * DH + a];
  }
  // backward solve L^T z = y
  #pragma unroll
  for (int a = DH - 1; a >= 0; --a) {
    double s = y[a];
    #pragma unroll
    for (int b = 0; b < DH; ++b)
      if (b > a) s = fma(-Li[b * DH + a], y[b], s);
    y[a] = s / Li[a * DH + a];
  }
  #pragma unroll
  for (int a = 0; a < DH; ++a) Z[((size_t)i * DH + a) * r + k] = y[a];
}

// ---------------------------------------------------------------------
// Fused tCG vector updates (guarded; coefficients read from ctrl):
//   snapshot: eta_snap[j] = eta, delta_snap[j] = delta  (pre-update)
//   eta += coef * delta;  if (!stop_pending) { r += alpha*Hd; rr+=r^2 }
// ---------------------------------------------------------------------
// Two elements per thread: the kernel is a pure fp64 stream (4 reads
// + 4 writes per element) and was memory-latency-bound at 1M-pose
// scale with one 8-byte access per lane per array.
template <int CF = CF_NONE>
__global__ void k_tcg_update(double* __restrict__ eta,
                             double* __restrict__ rvec,
                             const double* __restrict__ delta,
                             const double* __restrict__ Hd,
                             double* __restrict__ eta_snap,
                             double* __restrict__ delta_snap,
                             double* __restrict__ ctrl,
                             long total) {
  if (guarded_off(ctrl, ST_RUN)) return;
  const long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  const double coef = ctrl[C_COEF];
  const int stop_pending = (int)ctrl[C_STOP_PENDING];
  const int j = (int)ctrl[C_ITER];
  double rr = 0.0;
  #pragma unroll
  for (int u = 0; u < 4; ++u) {
    const long i = i0 + u;
    if (i < total) {
      const double dl = delta[i];
      // j == 0: eta starts the solve at 0 — write instead of
      // accumulate, so the eta buffer needs no per-solve zeroing pass
      const double et = (j == 0) ? 0.0 : eta[i];
      // snapshots are write-once, re-read only on a (rare) rejection
      // replay: nontemporal stores skip L2 write-allocate, halving
      // the kernel's cache pressure (dominant at 1M-pose scale)
      __builtin_nontemporal_store(et, &eta_snap[(size_t)j * total + i]);
      __builtin_nontemporal_store(dl, &delta_snap[(size_t)j * total + i]);
      eta[i] = fma(coef, dl, et);
      if (!stop_pending) {
        const double rn = fma(coef, Hd[i], rvec[i]);
        rvec[i] = rn;
        rr = fma(rn, rn, rr);
      }
    }
  }
  if (!stop_pending) block_reduce_atomic(rr, ctrl + C_DOT1);
  if (CF != CF_NONE) {
    if (fanin_last_block(ctrl) && threadIdx.x == 0)
      run_ctrl_tail(CF, ctrl, nullptr);
  }
}

// delta = -z + beta * delta
__global__ void k_tcg_delta(double* __restrict__ delta,
                            const double* __restrict__ z,
                            double* __restrict__ ctrl, long total) {
  if (guarded_off(ctrl, ST_RUN)) return;
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  delta[i] = fma(ctrl[C_BETA], delta[i], -z[i]);
}

// step = eta_snap[jstar] + tau*delta_snap[jstar], or current eta
__global__ void k_form_step(double* __restrict__ step,
                            const double* __restrict__ eta,
                            const double* __restrict__ eta_snap,
                            const double* __restrict__ delta_snap,
                            const double* __restrict__ ctrl, long total) {
  if (guarded_off(ctrl, ST_TCG_STOP)) return;
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  if (ctrl[C_USE_CURRENT] != 0.0) {
    step[i] = eta[i];
  } else {
    const int j = (int)ctrl[C_JSTAR];
    const double tau = ctrl[C_TAUSTAR];
    step[i] = fma(tau, delta_snap[(size_t)j * total + i],
                  eta_snap[(size_t)j * total + i]);
  }
}

// out = a*A + b*B elementwise (unguarded helper)
__global__ void k_axpby(const double* __restrict__ A,
                        const double* __restrict__ B,
                        double a, double b,
                        double* __restrict__ out, long total) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  double v = a * A[i];
  if (B) v = fma(b, B[i], v);
  out[i] = v;
}

// dots: ctrl[slot] += <A, B>, optionally ctrl[slot2] += <A, C>
template <int CF = CF_NONE>
__global__ void k_dots(const double* __restrict__ A,
                       const double* __restrict__ B,
                       const double* __restrict__ C,
                       double* __restrict__ ctrl,
                       int slot, int slot2, long total, int guard,
                       double* __restrict__ cf_out = nullptr) {
  if (guarded_off(ctrl, guard)) return;
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  double d0 = 0.0, d1 = 0.0;
  if (i < total) {
    const double a = A[i];
    d0 = a * B[i];
    if (C) d1 = a * C[i];
  }
  block_reduce_atomic(d0, ctrl + slot);
  if (C && slot2 >= 0) block_reduce_atomic(d1, ctrl + slot2);
  if (CF != CF_NONE) {
    if (fanin_last_block(ctrl) && threadIdx.x == 0)
      run_ctrl_tail(CF, ctrl, cf_out);
  }
}

// ---------------------------------------------------------------------
// Persistent solve kernel: the ENTIRE pre-sync local solve — G
// assembly, gradient phase, z0, the Steihaug tCG loop, the first
// candidate (replay + polar retraction + f evaluation) and the
// acceptance test — as ONE kernel with device-side grid barriers
// between stages, instead of ~60 dependent launches at the ~4 us
// dispatch-latency floor. Valid when every stage fits one resident
// grid (blocks <= 256 on 256 CUs => all workgroups co-resident, so
// the barrier cannot deadlock) and the dense preconditioner is in
// use. Reduction tails reuse the CF fan-in machinery; stage
// ordering/visibility comes from a generation-counter barrier with
// agent-scope fences. The control block is written to mapped pinned
// host memory by the last stage, so the host-side acceptance /
// shrink logic (solve_postsync) proceeds after one stream sync.
// ---------------------------------------------------------------------
__device__ __forceinline__ void grid_barrier(unsigned int* cnt,
                                             unsigned int* gen,
                                             unsigned int nb) {
  __syncthreads();
  __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
  if (threadIdx.x == 0) {
    const unsigned int g =
        __hip_atomic_load(gen, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    const unsigned int old = __hip_atomic_fetch_add(
        cnt, 1u, __ATOMIC_ACQ_REL, __HIP_MEMORY_SCOPE_AGENT);
    if (old == nb - 1) {
      __hip_atomic_store(cnt, 0u, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
      __hip_atomic_fetch_add(gen, 1u, __ATOMIC_RELEASE,
                             __HIP_MEMORY_SCOPE_AGENT);
    } else {
      long spins = 0;
      while (__hip_atomic_load(gen, __ATOMIC_ACQUIRE,
                               __HIP_MEMORY_SCOPE_AGENT) == g) {
        __builtin_amdgcn_s_sleep(8);
        if (++spins > 400000000L) __builtin_trap();  // lost-block guard
      }
    }
  }
  __syncthreads();
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
}

// k_ctrl_candidate's body as a device function (single thread).
__device__ void ctrl_candidate_logic(double* ctrl) {
  if (ctrl[C_STATUS] != (double)ST_TCG_STOP) return;
  const double radius = ctrl[C_RADIUS];
  const int hlen = (int)ctrl[C_HLEN];
  double dm = 0.0;
  int jstar = -1;
  double taustar = 0.0;
  for (int j = 0; j < hlen && j < MAX_TCG; ++j) {
    const double z_r = ctrl[H_ZR(j)];
    const double d_Hd = ctrl[H_DHD(j)];
    const double e_Pe = ctrl[H_EPE(j)];
    const double e_Pd = ctrl[H_EPD(j)];
    const double d_Pd = ctrl[H_DPD(j)];
    const double alpha = z_r / d_Hd;
    const double e_Pe_new = e_Pe + 2.0 * alpha * e_Pd + alpha * alpha * d_Pd;
    if (d_Hd <= 0.0 || e_Pe_new >= radius * radius) {
      const double disc = e_Pd * e_Pd + d_Pd * (radius * radius - e_Pe);
      const double tau = (d_Pd > 0.0)
          ? (-e_Pd + sqrt(fmax(disc, 0.0))) / d_Pd : 0.0;
      dm += tau * z_r - 0.5 * tau * tau * d_Hd;
      jstar = j;
      taustar = tau;
      break;
    }
    dm += 0.5 * alpha * z_r;
  }
  if (jstar < 0) {
    ctrl[C_USE_CURRENT] = 1.0;
  } else {
    ctrl[C_USE_CURRENT] = 0.0;
    ctrl[C_JSTAR] = (double)jstar;
    ctrl[C_TAUSTAR] = taustar;
  }
  ctrl[C_DM] = dm;
  ctrl[C_DOT0] = 0.0;
  ctrl[C_DOT2] = 0.0;
}

template <int D, int R>
__global__ void k_solve_persist(
    const int* __restrict__ q_rp, const int* __restrict__ q_ci,
    const double* __restrict__ q_vals, const double* __restrict__ X,
    const double* __restrict__ G,   // linear term (read; may be null)
    double* __restrict__ Gw,        // == G when assembling here, else null
    const double* __restrict__ gE0, const long* __restrict__ g_local_pose,
    const long* __restrict__ g_nbr_slot, const double* __restrict__ nbr,
    const double* __restrict__ g_w, int g_ne,
    const float* __restrict__ Minv, double* __restrict__ eta,
    double* __restrict__ rvec, double* __restrict__ delta,
    double* __restrict__ z, double* __restrict__ Hd,
    double* __restrict__ eta_snap, double* __restrict__ delta_snap,
    double* __restrict__ step, double* __restrict__ Xprop,
    double* __restrict__ ctrl, double* __restrict__ ctrl_host,
    unsigned int* __restrict__ gbar, int n, int N, long total,
    int max_inner, double tol, double Delta0, double theta, double kappa,
    double accept_rho, int jmul, int full) {
  constexpr int dh = D + 1;
  const int tid = blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const unsigned int nb = gridDim.x;
  unsigned int* cnt = gbar;
  unsigned int* gen = gbar + 1;
#define PSTAMP(i) \
  if (blockIdx.x == 0 && threadIdx.x == 0) \
    ctrl_host[CTRL_SIZE + (i)] = (double)wall_clock64();
  PSTAMP(0)

  if (full) {
  // --- S0: zero ctrl (+ G when assembling) --------------------------
  for (long t = tid; t < CTRL_SIZE; t += stride) ctrl[t] = 0.0;
  if (Gw) {
    for (long t = tid; t < total; t += stride) Gw[t] = 0.0;
    grid_barrier(cnt, gen, nb);
    // --- S0b: G assembly (edge-element stride loop, scatter-add) ----
    const long ge = (long)g_ne * dh * R;
    for (long i = tid; i < ge; i += stride) {
      const int e = (int)(i / (dh * R));
      const int t = (int)(i % (dh * R));
      const int c = t / R, k = t % R;
      const double* Ee = gE0 + (size_t)e * dh * dh;
      const double* Xn = nbr + (size_t)g_nbr_slot[e] * dh * R;
      double acc = 0.0;
      #pragma unroll
      for (int b = 0; b < dh; ++b)
        acc = fma(Ee[c * dh + b], Xn[b * R + k], acc);
      atomicAdd(&Gw[((size_t)g_local_pose[e] * dh + c) * R + k],
                -g_w[e] * acc);
    }
  }
  grid_barrier(cnt, gen, nb);
  PSTAMP(1)

  // --- S1: gradient phase (pose threads) + init tail ----------------
  {
    double d_fx = 0.0, d_gn = 0.0, d_gx = 0.0;
    if (tid < n) {
      double acc[dh][R];
      #pragma unroll
      for (int c = 0; c < dh; ++c)
        #pragma unroll
        for (int k = 0; k < R; ++k) acc[c][k] = 0.0;
      const int s0 = q_rp[tid], e0 = q_rp[tid + 1];
      for (int p = s0; p < e0; ++p) {
        const double* B = q_vals + (size_t)p * dh * dh;
        const double* Vj = X + (size_t)q_ci[p] * dh * R;
        #pragma unroll
        for (int c = 0; c < dh; ++c)
          #pragma unroll
          for (int cc = 0; cc < dh; ++cc) {
            const double b = B[c * dh + cc];
            #pragma unroll
            for (int k = 0; k < R; ++k)
              acc[c][k] = fma(b, Vj[cc * R + k], acc[c][k]);
          }
      }
      const double* Xi = X + (size_t)tid * dh * R;
      const double* Gi = G ? G + (size_t)tid * dh * R : nullptr;
      #pragma unroll
      for (int c = 0; c < dh; ++c)
        #pragma unroll
        for (int k = 0; k < R; ++k) {
          double g = Gi ? Gi[c * R + k] : 0.0;
          double v = acc[c][k] + g;
          acc[c][k] = v;
          d_fx = fma(v, Xi[c * R + k], d_fx);
          d_gx = fma(g, Xi[c * R + k], d_gx);
        }
      // tangent projection at X
      double S[D][D];
      #pragma unroll
      for (int a = 0; a < D; ++a)
        #pragma unroll
        for (int b = 0; b < D; ++b) {
          double sv = 0.0;
          #pragma unroll
          for (int k = 0; k < R; ++k)
            sv = fma(Xi[a * R + k], acc[b][k], sv);
          S[a][b] = sv;
        }
      #pragma unroll
      for (int a = 0; a < D; ++a)
        #pragma unroll
        for (int b = a; b < D; ++b) {
          const double sv = 0.5 * (S[a][b] + S[b][a]);
          S[a][b] = sv;
          S[b][a] = sv;
        }
      #pragma unroll
      for (int a = 0; a < D; ++a)
        #pragma unroll
        for (int k = 0; k < R; ++k) {
          double v = acc[a][k];
          #pragma unroll
          for (int b = 0; b < D; ++b)
            v = fma(-S[a][b], Xi[b * R + k], v);
          acc[a][k] = v;
        }
      double* Ri = rvec + (size_t)tid * dh * R;
      #pragma unroll
      for (int c = 0; c < dh; ++c)
        #pragma unroll
        for (int k = 0; k < R; ++k) {
          Ri[c * R + k] = acc[c][k];
          d_gn = fma(acc[c][k], acc[c][k], d_gn);
        }
    }
    if (tid < total) z[tid] = 0.0;  // j-split precond accumulates
    block_reduce_atomic(d_fx, ctrl + C_DOT0);
    block_reduce_atomic(d_gn, ctrl + C_DOT1);
    block_reduce_atomic(d_gx, ctrl + C_DOT2);
    if (fanin_last_block(ctrl) && threadIdx.x == 0) {
      // k_ctrl_init body
      const double fX = 0.5 * (ctrl_load(ctrl + C_DOT0)
                               + ctrl_load(ctrl + C_DOT2));
      const double gn0sq = ctrl_load(ctrl + C_DOT1);
      ctrl_store(ctrl + C_FX, fX);
      ctrl_store(ctrl + C_GN0SQ, gn0sq);
      ctrl_store(ctrl + C_RR, gn0sq);
      const double norm_r0 = sqrt(gn0sq);
      ctrl_store(ctrl + C_BOUND, norm_r0 * fmin(pow(norm_r0, theta), kappa));
      ctrl_store(ctrl + C_RADIUS, Delta0);
      ctrl_store(ctrl + C_EPE, 0.0);
      ctrl_store(ctrl + C_EPD, 0.0);
      ctrl_store(ctrl + C_ITER, 0.0);
      ctrl_store(ctrl + C_J, 0.0);
      ctrl_store(ctrl + C_USE_CURRENT, 0.0);
      ctrl_store(ctrl + C_STOP_PENDING, 0.0);
      ctrl_store(ctrl + C_BETA, 0.0);
      ctrl_store(ctrl + C_STATUS, (norm_r0 < tol) ? (double)ST_NO_UPDATE
                                                  : (double)ST_RUN);
      ctrl_store(ctrl + C_DOT0, 0.0);
      ctrl_store(ctrl + C_DOT1, 0.0);
      ctrl_store(ctrl + C_DOT2, 0.0);
      ctrl_store(ctrl + C_DOT3, 0.0);
    }
  }
  grid_barrier(cnt, gen, nb);
  }  // full: assembly + gradient phase
  PSTAMP(2)

  if (full && ctrl[C_STATUS] == (double)ST_RUN) {
    // --- S2: z0 = P_X(Minv r0), <z0,r0> -> tail_z0; delta0 = -z0 ----
    // j-split dense apply: the extra (jmul-1)*total threads of the
    // widened grid each take a chunk of the j-range (the apply is
    // latency-bound on the L2-cold pass over Minv, so memory-level
    // parallelism is the lever); 4 accumulator chains per thread.
    if (tid < (long)total * jmul) {
      const long e = tid % total;
      const int q = (int)(tid / total);
      const int i = (int)(e / R), k = (int)(e % R);
      const int jchunk = (N + jmul - 1) / jmul;
      const int jlo = q * jchunk, jhi = min(N, jlo + jchunk);
      double a0 = 0.0, a1 = 0.0, a2 = 0.0, a3 = 0.0;
      int jj = jlo;
      for (; jj + 4 <= jhi; jj += 4) {
        a0 = fma((double)Minv[(size_t)(jj + 0) * N + i],
                 rvec[(size_t)(jj + 0) * R + k], a0);
        a1 = fma((double)Minv[(size_t)(jj + 1) * N + i],
                 rvec[(size_t)(jj + 1) * R + k], a1);
        a2 = fma((double)Minv[(size_t)(jj + 2) * N + i],
                 rvec[(size_t)(jj + 2) * R + k], a2);
        a3 = fma((double)Minv[(size_t)(jj + 3) * N + i],
                 rvec[(size_t)(jj + 3) * R + k], a3);
      }
      for (; jj < jhi; ++jj)
        a0 = fma((double)Minv[(size_t)jj * N + i],
                 rvec[(size_t)jj * R + k], a0);
      const double part = (a0 + a1) + (a2 + a3);
      if (jmul == 1) z[e] = part;
      else atomicAdd(&z[e], part);
    }
    grid_barrier(cnt, gen, nb);
    {
      double d0 = 0.0;
      if (tid < n) {
        const double* Xi = X + (size_t)tid * dh * R;
        double* Zi = z + (size_t)tid * dh * R;
        const double* Ri = rvec + (size_t)tid * dh * R;
        double Vt[dh][R];
        #pragma unroll
        for (int c = 0; c < dh; ++c)
          #pragma unroll
          for (int k = 0; k < R; ++k) Vt[c][k] = Zi[c * R + k];
        double S[D][D];
        #pragma unroll
        for (int a = 0; a < D; ++a)
          #pragma unroll
          for (int b = 0; b < D; ++b) {
            double sv = 0.0;
            #pragma unroll
            for (int k = 0; k < R; ++k) sv = fma(Xi[a * R + k], Vt[b][k], sv);
            S[a][b] = sv;
          }
        #pragma unroll
        for (int a = 0; a < D; ++a)
          #pragma unroll
          for (int b = a; b < D; ++b) {
            const double sv = 0.5 * (S[a][b] + S[b][a]);
            S[a][b] = sv;
            S[b][a] = sv;
          }
        #pragma unroll
        for (int a = 0; a < D; ++a)
          #pragma unroll
          for (int k = 0; k < R; ++k) {
            double v = Vt[a][k];
            #pragma unroll
            for (int b = 0; b < D; ++b)
              v = fma(-S[a][b], Xi[b * R + k], v);
            Vt[a][k] = v;
          }
        #pragma unroll
        for (int c = 0; c < dh; ++c)
          #pragma unroll
          for (int k = 0; k < R; ++k) {
            Zi[c * R + k] = Vt[c][k];
            d0 = fma(Vt[c][k], Ri[c * R + k], d0);
          }
      }
      block_reduce_atomic(d0, ctrl + C_DOT0);
      if (fanin_last_block(ctrl) && threadIdx.x == 0)
        ctrl_tail_z0(ctrl);
    }
    grid_barrier(cnt, gen, nb);
    if (tid < total) delta[tid] = -z[tid];
  }  // full: z0 phase
  PSTAMP(3)

  {
    // --- S3: the tCG loop -------------------------------------------
    for (int it = 0; it < max_inner; ++it) {
      grid_barrier(cnt, gen, nb);  // prior delta writes visible
      if (ctrl[C_STATUS] != (double)ST_RUN) break;

      // H: Hd = P_X(Q delta), d_Qd -> C_DOT0, tail_alpha
      {
        double d0 = 0.0;
        if (tid < n) {
          double acc[dh][R];
          #pragma unroll
          for (int c = 0; c < dh; ++c)
            #pragma unroll
            for (int k = 0; k < R; ++k) acc[c][k] = 0.0;
          const int s0 = q_rp[tid], e0 = q_rp[tid + 1];
          for (int p = s0; p < e0; ++p) {
            const double* B = q_vals + (size_t)p * dh * dh;
            const double* Vj = delta + (size_t)q_ci[p] * dh * R;
            #pragma unroll
            for (int c = 0; c < dh; ++c)
              #pragma unroll
              for (int cc = 0; cc < dh; ++cc) {
                const double b = B[c * dh + cc];
                #pragma unroll
                for (int k = 0; k < R; ++k)
                  acc[c][k] = fma(b, Vj[cc * R + k], acc[c][k]);
              }
          }
          const double* Xi = X + (size_t)tid * dh * R;
          double S[D][D];
          #pragma unroll
          for (int a = 0; a < D; ++a)
            #pragma unroll
            for (int b = 0; b < D; ++b) {
              double sv = 0.0;
              #pragma unroll
              for (int k = 0; k < R; ++k)
                sv = fma(Xi[a * R + k], acc[b][k], sv);
              S[a][b] = sv;
            }
          #pragma unroll
          for (int a = 0; a < D; ++a)
            #pragma unroll
            for (int b = a; b < D; ++b) {
              const double sv = 0.5 * (S[a][b] + S[b][a]);
              S[a][b] = sv;
              S[b][a] = sv;
            }
          #pragma unroll
          for (int a = 0; a < D; ++a)
            #pragma unroll
            for (int k = 0; k < R; ++k) {
              double v = acc[a][k];
              #pragma unroll
              for (int b = 0; b < D; ++b)
                v = fma(-S[a][b], Xi[b * R + k], v);
              acc[a][k] = v;
            }
          double* Oi = Hd + (size_t)tid * dh * R;
          const double* Wi = delta + (size_t)tid * dh * R;
          #pragma unroll
          for (int c = 0; c < dh; ++c)
            #pragma unroll
            for (int k = 0; k < R; ++k) {
              Oi[c * R + k] = acc[c][k];
              d0 = fma(acc[c][k], Wi[c * R + k], d0);
            }
        }
        block_reduce_atomic(d0, ctrl + C_DOT0);
        if (fanin_last_block(ctrl) && threadIdx.x == 0)
          ctrl_tail_alpha(ctrl);
      }
      grid_barrier(cnt, gen, nb);

      // U: eta/r update + snapshots, rr -> C_DOT1, tail_rr
      {
        const double coef = ctrl[C_COEF];
        const int stop_pending = (int)ctrl[C_STOP_PENDING];
        const int j = (int)ctrl[C_ITER];
        double rr = 0.0;
        if (tid < total) z[tid] = 0.0;  // next j-split apply accumulates
        if (tid < total) {
          const double dl = delta[tid];
          const double et = (j == 0) ? 0.0 : eta[tid];
          __builtin_nontemporal_store(et,
              &eta_snap[(size_t)j * total + tid]);
          __builtin_nontemporal_store(dl,
              &delta_snap[(size_t)j * total + tid]);
          eta[tid] = fma(coef, dl, et);
          if (!stop_pending) {
            const double rn = fma(coef, Hd[tid], rvec[tid]);
            rvec[tid] = rn;
            rr = rn * rn;
          }
        }
        if (!stop_pending) block_reduce_atomic(rr, ctrl + C_DOT1);
        if (fanin_last_block(ctrl) && threadIdx.x == 0)
          ctrl_tail_rr(ctrl);
      }
      grid_barrier(cnt, gen, nb);
      if (ctrl[C_STATUS] != (double)ST_RUN) break;

      // P: z = Minv r (j-split dense fp32 apply; see z0 stage)
      if (tid < (long)total * jmul) {
        const long e = tid % total;
        const int q = (int)(tid / total);
        const int i = (int)(e / R), k = (int)(e % R);
        const int jchunk = (N + jmul - 1) / jmul;
        const int jlo = q * jchunk, jhi = min(N, jlo + jchunk);
        double a0 = 0.0, a1 = 0.0, a2 = 0.0, a3 = 0.0;
        int jj = jlo;
        for (; jj + 4 <= jhi; jj += 4) {
          a0 = fma((double)Minv[(size_t)(jj + 0) * N + i],
                   rvec[(size_t)(jj + 0) * R + k], a0);
          a1 = fma((double)Minv[(size_t)(jj + 1) * N + i],
                   rvec[(size_t)(jj + 1) * R + k], a1);
          a2 = fma((double)Minv[(size_t)(jj + 2) * N + i],
                   rvec[(size_t)(jj + 2) * R + k], a2);
          a3 = fma((double)Minv[(size_t)(jj + 3) * N + i],
                   rvec[(size_t)(jj + 3) * R + k], a3);
        }
        for (; jj < jhi; ++jj)
          a0 = fma((double)Minv[(size_t)jj * N + i],
                   rvec[(size_t)jj * R + k], a0);
        const double part = (a0 + a1) + (a2 + a3);
        if (jmul == 1) z[e] = part;
        else atomicAdd(&z[e], part);
      }
      grid_barrier(cnt, gen, nb);

      // J: project z at X, <z, r> -> C_DOT0, tail_beta
      {
        double d0 = 0.0;
        if (tid < n) {
          const double* Xi = X + (size_t)tid * dh * R;
          double* Zi = z + (size_t)tid * dh * R;
          const double* Ri = rvec + (size_t)tid * dh * R;
          double Vt[dh][R];
          #pragma unroll
          for (int c = 0; c < dh; ++c)
            #pragma unroll
            for (int k = 0; k < R; ++k) Vt[c][k] = Zi[c * R + k];
          double S[D][D];
          #pragma unroll
          for (int a = 0; a < D; ++a)
            #pragma unroll
            for (int b = 0; b < D; ++b) {
              double sv = 0.0;
              #pragma unroll
              for (int k = 0; k < R; ++k)
                sv = fma(Xi[a * R + k], Vt[b][k], sv);
              S[a][b] = sv;
            }
          #pragma unroll
          for (int a = 0; a < D; ++a)
            #pragma unroll
            for (int b = a; b < D; ++b) {
              const double sv = 0.5 * (S[a][b] + S[b][a]);
              S[a][b] = sv;
              S[b][a] = sv;
            }
          #pragma unroll
          for (int a = 0; a < D; ++a)
            #pragma unroll
            for (int k = 0; k < R; ++k) {
              double v = Vt[a][k];
              #pragma unroll
              for (int b = 0; b < D; ++b)
                v = fma(-S[a][b], Xi[b * R + k], v);
              Vt[a][k] = v;
            }
          #pragma unroll
          for (int c = 0; c < dh; ++c)
            #pragma unroll
            for (int k = 0; k < R; ++k) {
              Zi[c * R + k] = Vt[c][k];
              d0 = fma(Vt[c][k], Ri[c * R + k], d0);
            }
        }
        block_reduce_atomic(d0, ctrl + C_DOT0);
        if (fanin_last_block(ctrl) && threadIdx.x == 0)
          ctrl_tail_beta(ctrl);
      }
      grid_barrier(cnt, gen, nb);

      // D: delta = beta*delta - z
      if (tid < total)
        delta[tid] = fma(ctrl[C_BETA], delta[tid], -z[tid]);
    }
    grid_barrier(cnt, gen, nb);
    PSTAMP(4)
    if (full && blockIdx.x == 0 && threadIdx.x == 0) {
      // k_ctrl_tcg_end body
      if (ctrl[C_STATUS] == (double)ST_RUN) {
        ctrl_store(ctrl + C_STATUS, (double)ST_TCG_STOP);
        ctrl_store(ctrl + C_J, ctrl[C_ITER]);
        ctrl_store(ctrl + C_HLEN, ctrl[C_ITER]);
        ctrl_store(ctrl + C_USE_CURRENT, 1.0);
      }
      ctrl_candidate_logic(ctrl);
    }
    grid_barrier(cnt, gen, nb);
  }

  // --- S4: first candidate: form step, retract, evaluate, accept ----
  PSTAMP(5)
  if (full && ctrl[C_STATUS] == (double)ST_TCG_STOP) {
    if (tid < total) {
      if (ctrl[C_USE_CURRENT] != 0.0) {
        step[tid] = eta[tid];
      } else {
        const int j = (int)ctrl[C_JSTAR];
        const double tau = ctrl[C_TAUSTAR];
        step[tid] = fma(tau, delta_snap[(size_t)j * total + tid],
                        eta_snap[(size_t)j * total + tid]);
      }
    }
    grid_barrier(cnt, gen, nb);
    // polar retraction Xprop = polar(X + step)
    if (tid < n) {
      double Mt[dh][R];
      const double* Ai = X + (size_t)tid * dh * R;
      const double* Bi = step + (size_t)tid * dh * R;
      #pragma unroll
      for (int c = 0; c < dh; ++c)
        #pragma unroll
        for (int k = 0; k < R; ++k) Mt[c][k] = Ai[c * R + k] + Bi[c * R + k];
      double S[D][D];
      #pragma unroll
      for (int a = 0; a < D; ++a)
        #pragma unroll
        for (int b = 0; b < D; ++b) {
          double s = 0.0;
          #pragma unroll
          for (int k = 0; k < R; ++k) s = fma(Mt[a][k], Mt[b][k], s);
          S[a][b] = s;
        }
      double Gi2[D][D];
      spd_inv_sqrt<D>(S, Gi2);
      double* Oi = Xprop + (size_t)tid * dh * R;
      #pragma unroll
      for (int a = 0; a < D; ++a)
        #pragma unroll
        for (int k = 0; k < R; ++k) {
          double s = 0.0;
          #pragma unroll
          for (int b = 0; b < D; ++b) s = fma(Gi2[a][b], Mt[b][k], s);
          Oi[a * R + k] = s;
        }
      #pragma unroll
      for (int k = 0; k < R; ++k) Oi[D * R + k] = Mt[D][k];
    }
    grid_barrier(cnt, gen, nb);
    // f(Xprop) dots: <Q Xprop, Xprop> -> C_DOT0, <G, Xprop> -> C_DOT2
    {
      double d0 = 0.0, d1 = 0.0;
      if (tid < n) {
        double acc[dh][R];
        #pragma unroll
        for (int c = 0; c < dh; ++c)
          #pragma unroll
          for (int k = 0; k < R; ++k) acc[c][k] = 0.0;
        const int s0 = q_rp[tid], e0 = q_rp[tid + 1];
        for (int p = s0; p < e0; ++p) {
          const double* B = q_vals + (size_t)p * dh * dh;
          const double* Vj = Xprop + (size_t)q_ci[p] * dh * R;
          #pragma unroll
          for (int c = 0; c < dh; ++c)
            #pragma unroll
            for (int cc = 0; cc < dh; ++cc) {
              const double b = B[c * dh + cc];
              #pragma unroll
              for (int k = 0; k < R; ++k)
                acc[c][k] = fma(b, Vj[cc * R + k], acc[c][k]);
            }
        }
        const double* Vi = Xprop + (size_t)tid * dh * R;
        const double* Gi = G ? G + (size_t)tid * dh * R : nullptr;
        #pragma unroll
        for (int c = 0; c < dh; ++c)
          #pragma unroll
          for (int k = 0; k < R; ++k) {
            const double g = Gi ? Gi[c * R + k] : 0.0;
            d0 = fma(acc[c][k], Vi[c * R + k], d0);
            d1 = fma(g, Vi[c * R + k], d1);
          }
      }
      block_reduce_atomic(d0, ctrl + C_DOT0);
      block_reduce_atomic(d1, ctrl + C_DOT2);
      if (fanin_last_block(ctrl) && threadIdx.x == 0) {
        // k_ctrl_accept body
        if (ctrl[C_STATUS] == (double)ST_TCG_STOP) {
          const double fprop = 0.5 * ctrl_load(ctrl + C_DOT0)
                               + ctrl_load(ctrl + C_DOT2);
          ctrl_store(ctrl + C_DOT0, 0.0);
          ctrl_store(ctrl + C_DOT2, 0.0);
          ctrl_store(ctrl + C_FPROP, fprop);
          const double fX = ctrl[C_FX];
          const double dm = ctrl[C_DM];
          const double rho = (fX - fprop) / fmax(dm, 1e-300);
          ctrl_store(ctrl + C_RHO, rho);
          if (rho > accept_rho && fprop <= fX)
            ctrl_store(ctrl + C_STATUS, (double)ST_ACCEPTED);
        }
      }
    }
  }
  grid_barrier(cnt, gen, nb);
  PSTAMP(6)

  // --- S5: publish the control block to the host --------------------
  if (full && blockIdx.x == 0)
    for (int t = threadIdx.x; t < CTRL_SIZE; t += blockDim.x)
      ctrl_host[t] = ctrl_load(ctrl + t);
  PSTAMP(7)
#undef PSTAMP
}

// ---------------------------------------------------------------------
// Single-wave tCG control kernels (device-side branch logic)
// ---------------------------------------------------------------------
__global__ void k_ctrl_init(double* ctrl, double tol, double Delta0,
                            double theta, double kappa) {
  if (threadIdx.x != 0) return;
  // C_DOT0 = <QX + G, X>, C_DOT2 = <G, X>
  // => f = 0.5 <QX, X> + <G, X> = 0.5 (C_DOT0 + C_DOT2)
  const double fX = 0.5 * (ctrl[C_DOT0] + ctrl[C_DOT2]);
  const double gn0sq = ctrl[C_DOT1];
  ctrl[C_FX] = fX;
  ctrl[C_GN0SQ] = gn0sq;
  ctrl[C_RR] = gn0sq;
  const double norm_r0 = sqrt(gn0sq);
  ctrl[C_BOUND] = norm_r0 * fmin(pow(norm_r0, theta), kappa);
  ctrl[C_RADIUS] = Delta0;
  ctrl[C_EPE] = 0.0;
  ctrl[C_EPD] = 0.0;
  ctrl[C_ITER] = 0.0;
  ctrl[C_J] = 0.0;
  ctrl[C_USE_CURRENT] = 0.0;
  ctrl[C_STOP_PENDING] = 0.0;
  ctrl[C_BETA] = 0.0;
  ctrl[C_STATUS] = (norm_r0 < tol) ? (double)ST_NO_UPDATE : (double)ST_RUN;
  ctrl[C_DOT0] = 0.0;
  ctrl[C_DOT1] = 0.0;
  ctrl[C_DOT2] = 0.0;
  ctrl[C_DOT3] = 0.0;
}

// after z0 projection+dot (C_DOT0 = <z0, r0>)
__global__ void k_ctrl_z0(double* ctrl) {
  if (threadIdx.x != 0) return;
  if (ctrl[C_STATUS] != (double)ST_RUN) return;
  const double zr = ctrl[C_DOT0];
  ctrl[C_ZR] = zr;
  ctrl[C_DPD] = zr;
  ctrl[C_DOT0] = 0.0;
}

// after Hd projection+dot (C_DOT0 = <delta, Hd>): alpha / boundary logic
__global__ void k_ctrl_alpha(double* ctrl) {
  if (threadIdx.x != 0) return;
  if (ctrl[C_STATUS] != (double)ST_RUN) return;
  const int j = (int)ctrl[C_ITER];
  const double d_Hd = ctrl[C_DOT0];
  ctrl[C_DOT0] = 0.0;
  const double z_r = ctrl[C_ZR];
  const double e_Pe = ctrl[C_EPE], e_Pd = ctrl[C_EPD], d_Pd = ctrl[C_DPD];
  const double radius = ctrl[C_RADIUS];
  // scalar history for radius-replay (shrink loop reuses the Krylov path)
  ctrl[H_ZR(j)] = z_r;
  ctrl[H_DHD(j)] = d_Hd;
  ctrl[H_EPE(j)] = e_Pe;
  ctrl[H_EPD(j)] = e_Pd;
  ctrl[H_DPD(j)] = d_Pd;
  const double alpha = z_r / d_Hd;
  const double e_Pe_new = e_Pe + 2.0 * alpha * e_Pd + alpha * alpha * d_Pd;
  if (d_Hd <= 0.0 || e_Pe_new >= radius * radius) {
    const double disc = e_Pd * e_Pd + d_Pd * (radius * radius - e_Pe);
    const double tau = (d_Pd > 0.0)
        ? (-e_Pd + sqrt(fmax(disc, 0.0))) / d_Pd : 0.0;
    ctrl[C_COEF] = tau;
    ctrl[C_STOP_PENDING] = 1.0;
  } else {
    ctrl[C_COEF] = alpha;
    ctrl[C_EPE] = e_Pe_new;
    ctrl[C_STOP_PENDING] = 0.0;
  }
}

// after k_tcg_update (C_DOT1 = ||r||^2 when continuing)
__global__ void k_ctrl_rr(double* ctrl) {
  if (threadIdx.x != 0) return;
  if (ctrl[C_STATUS] != (double)ST_RUN) return;
  const int j = (int)ctrl[C_ITER];
  if (ctrl[C_STOP_PENDING] != 0.0) {
    ctrl[C_STATUS] = (double)ST_TCG_STOP;
    ctrl[C_J] = (double)j;         // truncated at snapshot j
    ctrl[C_HLEN] = (double)(j + 1);
    ctrl[C_USE_CURRENT] = 0.0;
    return;
  }
  const double rr = ctrl[C_DOT1];
  ctrl[C_DOT1] = 0.0;
  ctrl[C_RR] = rr;
  if (sqrt(rr) <= ctrl[C_BOUND]) {
    ctrl[C_STATUS] = (double)ST_TCG_STOP;
    ctrl[C_J] = (double)(j + 1);
    ctrl[C_HLEN] = (double)(j + 1);
    ctrl[C_USE_CURRENT] = 1.0;     // converged: step = current eta
  }
}

// after z projection+dot (C_DOT0 = <z, r>): beta and delta scalars
__global__ void k_ctrl_beta(double* ctrl) {
  if (threadIdx.x != 0) return;
  if (ctrl[C_STATUS] != (double)ST_RUN) return;
  const double z_r_new = ctrl[C_DOT0];
  ctrl[C_DOT0] = 0.0;
  const double z_r = ctrl[C_ZR];
  const double alpha = ctrl[C_COEF];
  const double beta = z_r_new / z_r;
  ctrl[C_BETA] = beta;
  ctrl[C_EPD] = beta * (ctrl[C_EPD] + alpha * ctrl[C_DPD]);
  ctrl[C_DPD] = z_r_new + beta * beta * ctrl[C_DPD];
  ctrl[C_ZR] = z_r_new;
  ctrl[C_ITER] += 1.0;
}

// end of the unrolled loop: cap at max_inner
__global__ void k_ctrl_tcg_end(double* ctrl) {
  if (threadIdx.x != 0) return;
  if (ctrl[C_STATUS] != (double)ST_RUN) return;
  ctrl[C_STATUS] = (double)ST_TCG_STOP;
  ctrl[C_J] = ctrl[C_ITER];
  ctrl[C_HLEN] = ctrl[C_ITER];
  ctrl[C_USE_CURRENT] = 1.0;
}

// candidate selection for the current radius: replay the stored Krylov
// scalars, find the truncation point and the model decrease.
__global__ void k_ctrl_candidate(double* ctrl) {
  if (threadIdx.x != 0) return;
  if (ctrl[C_STATUS] != (double)ST_TCG_STOP) return;
  const double radius = ctrl[C_RADIUS];
  const int hlen = (int)ctrl[C_HLEN];
  double dm = 0.0;
  int jstar = -1;
  double taustar = 0.0;
  for (int j = 0; j < hlen && j < MAX_TCG; ++j) {
    const double z_r = ctrl[H_ZR(j)];
    const double d_Hd = ctrl[H_DHD(j)];
    const double e_Pe = ctrl[H_EPE(j)];
    const double e_Pd = ctrl[H_EPD(j)];
    const double d_Pd = ctrl[H_DPD(j)];
    const double alpha = z_r / d_Hd;
    const double e_Pe_new = e_Pe + 2.0 * alpha * e_Pd + alpha * alpha * d_Pd;
    if (d_Hd <= 0.0 || e_Pe_new >= radius * radius) {
      const double disc = e_Pd * e_Pd + d_Pd * (radius * radius - e_Pe);
      const double tau = (d_Pd > 0.0)
          ? (-e_Pd + sqrt(fmax(disc, 0.0))) / d_Pd : 0.0;
      dm += tau * z_r - 0.5 * tau * tau * d_Hd;
      jstar = j;
      taustar = tau;
      break;
    }
    dm += 0.5 * alpha * z_r;
  }
  if (jstar < 0) {
    // full step (no truncation at this radius)
    ctrl[C_USE_CURRENT] = 1.0;
  } else {
    ctrl[C_USE_CURRENT] = 0.0;
    ctrl[C_JSTAR] = (double)jstar;
    ctrl[C_TAUSTAR] = taustar;
  }
  ctrl[C_DM] = dm;
  ctrl[C_DOT0] = 0.0;
  ctrl[C_DOT2] = 0.0;
}

// acceptance test after f(X_prop) dots (C_DOT0 = <QXp, Xp>, C_DOT2 = <G, Xp>)
__global__ void k_ctrl_accept(double* ctrl, double accept_rho) {
  if (threadIdx.x != 0) return;
  if (ctrl[C_STATUS] != (double)ST_TCG_STOP) return;
  const double fprop = 0.5 * ctrl[C_DOT0] + ctrl[C_DOT2];
  ctrl[C_DOT0] = 0.0;
  ctrl[C_DOT2] = 0.0;
  ctrl[C_FPROP] = fprop;
  const double fX = ctrl[C_FX];
  const double dm = ctrl[C_DM];
  const double rho = (fX - fprop) / fmax(dm, 1e-300);
  ctrl[C_RHO] = rho;
  if (rho > accept_rho && fprop <= fX) {
    ctrl[C_STATUS] = (double)ST_ACCEPTED;
  }
}

// shrink the radius after a rejection (host re-enqueues the candidate set)
__global__ void k_ctrl_shrink(double* ctrl) {
  if (threadIdx.x != 0) return;
  if (ctrl[C_STATUS] != (double)ST_TCG_STOP) return;
  ctrl[C_RADIUS] *= 0.25;
  ctrl[C_SHRINKS] += 1.0;
}

// ---------------------------------------------------------------------
// Assembly kernels
// ---------------------------------------------------------------------
// Q values: vals[slot] += w[edge_of[c]] * blocks[c]  (scatter-add)
__global__ void k_q_assemble(double* __restrict__ vals,
                             const double* __restrict__ blocks,
                             const long* __restrict__ slots,
                             const long* __restrict__ edge_of,
                             const double* __restrict__ w,
                             int ncontrib, int bsz) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= (long)ncontrib * bsz) return;
  const int c = i / bsz, e = i % bsz;
  atomicAdd(&vals[slots[c] * bsz + e], w[edge_of[c]] * blocks[i]);
}

// G values: Gt[local_pose[e]] += -w[e] * E0[e] @ nbr[slot[e]]
// one thread per (edge, c, k) output element
__global__ void k_g_assemble(double* __restrict__ Gt,
                             const double* __restrict__ E0,
                             const long* __restrict__ local_pose,
                             const long* __restrict__ nbr_slot,
                             const double* __restrict__ nbr,
                             const double* __restrict__ w,
                             int ne, int dh, int r) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= (long)ne * dh * r) return;
  const int e = i / (dh * r);
  const int t = i % (dh * r);
  const int c = t / r, k = t % r;
  const double* Ee = E0 + (size_t)e * dh * dh;
  const double* Xn = nbr + (size_t)nbr_slot[e] * dh * r;
  double acc = 0.0;
  for (int b = 0; b < dh; ++b) acc = fma(Ee[c * dh + b], Xn[b * r + k], acc);
  atomicAdd(&Gt[((size_t)local_pose[e] * dh + c) * r + k], -w[e] * acc);
}

// ---------------------------------------------------------------------
// GNC-TLS loop-closure re-weighting (reference PGOAgent::
// updateLoopClosuresWeights, PGOAgent.cpp:1181-1245, weight formula
// DPGO_robust.cpp:49-62). One thread per edge; endpoints are rows of X
// (local poses) or of the packed neighbor buffer (shared edges).
// Residual on the LIFTED variables: r^2 = kappa ||R^T Y1t - Y2t||_F^2
//                                      + tau ||p2 - p1 - t^T Y1t||^2.
// update_mask == 0 keeps the current weight (known inliers, non-owned
// shared edges under the owner-computes rule).
// ---------------------------------------------------------------------
template <int D, int R>
__global__ void k_gnc_weights(const double* __restrict__ X,
                              const double* __restrict__ nbr,
                              const long* __restrict__ e1_idx,
                              const unsigned char* __restrict__ e1_nbr,
                              const long* __restrict__ e2_idx,
                              const unsigned char* __restrict__ e2_nbr,
                              const double* __restrict__ Rm,
                              const double* __restrict__ tm,
                              const double* __restrict__ kappa,
                              const double* __restrict__ tau,
                              const unsigned char* __restrict__ update_mask,
                              const long* __restrict__ widx,
                              double* __restrict__ weights,
                              int ne, double mu, double barc_sq) {
  const int e = blockIdx.x * blockDim.x + threadIdx.x;
  if (e >= ne) return;
  if (!update_mask[e]) return;
  constexpr int dh = D + 1;
  const double* P1 = (e1_nbr[e] ? nbr : X) + (size_t)e1_idx[e] * dh * R;
  const double* P2 = (e2_nbr[e] ? nbr : X) + (size_t)e2_idx[e] * dh * R;
  const double* Re = Rm + (size_t)e * D * D;
  const double* te = tm + (size_t)e * D;
  double rot_err = 0.0, tran_err = 0.0;
  #pragma unroll
  for (int k = 0; k < R; ++k) {
    #pragma unroll
    for (int b = 0; b < D; ++b) {
      // (Y1 R)^T row b = sum_a R[a][b] * Y1t[a][:]
      double v = 0.0;
      #pragma unroll
      for (int a = 0; a < D; ++a) v = fma(Re[a * D + b], P1[a * R + k], v);
      const double diff = v - P2[b * R + k];
      rot_err = fma(diff, diff, rot_err);
    }
    double tv = P2[D * R + k] - P1[D * R + k];
    #pragma unroll
    for (int a = 0; a < D; ++a) tv = fma(-te[a], P1[a * R + k], tv);
    tran_err = fma(tv, tv, tran_err);
  }
  const double r_sq = kappa[e] * rot_err + tau[e] * tran_err;
  // GNC-TLS weight (eq. 14)
  const double upper = (mu + 1.0) / mu * barc_sq;
  const double lower = mu / (mu + 1.0) * barc_sq;
  double w;
  if (r_sq >= upper) w = 0.0;
  else if (r_sq <= lower) w = 1.0;
  else w = sqrt(barc_sq * mu * (mu + 1.0) / r_sq) - mu;
  weights[widx[e]] = w;
}

// ---------------------------------------------------------------------
// (d, r) dispatch: the supported compile-time shapes. SE(2): d=2,
// r in 2..6; SE(3): d=3, r in 3..8. Everything the reference exercises
// (r=5 RBCD, r=d batch) is covered; exotic shapes abort loudly.
// ---------------------------------------------------------------------
#define DPO_FOREACH_DR(F) \
  F(2, 2) F(2, 3) F(2, 4) F(2, 5) F(2, 6) \
  F(3, 3) F(3, 4) F(3, 5) F(3, 6) F(3, 7) F(3, 8)

static inline int blocks_for(long total, int bs) {
  return (int)((total + bs - 1) / bs);
}

static void dpo_bad_shape(int d, int r) {
  fprintf(stderr, "dpo_ops: unsupported (d=%d, r=%d) kernel shape\n", d, r);
  abort();
}

static void launch_spmm(const int* rp, const int* ci, const double* vals,
                        int n, int d, int r, const double* X, double* out,
                        const double* ctrl, int guard, hipStream_t s) {
  const int dh = d + 1;
  const int tile = dh * r;
  const int grid = blocks_for(n, 256 / tile);
#define CASE_SPMM(D, R) \
  if (d == D && r == R) { \
    hipLaunchKernelGGL((k_bsr_spmm<D + 1, R>), dim3(grid), dim3(256), 0, s, \
                       rp, ci, vals, X, out, n, ctrl, guard); \
    return; \
  }
  DPO_FOREACH_DR(CASE_SPMM)
#undef CASE_SPMM
  hipLaunchKernelGGL(k_bsr_spmm_gen, dim3(grid), dim3(256), 0, s,
                     rp, ci, vals, X, out, n, dh, r, ctrl, guard);
}

static inline bool hess_use_wide(int n);

template <int CF>
static void launch_proj_dots_cf(const double* X, const double* V,
                                const double* G, double* out,
                                const double* dotWith, double* ctrl,
                                int n, int d, int r, int dot_slot,
                                int dot_slot2, int guard, hipStream_t s) {
  const int grid = blocks_for(n, 256);
  const bool wide = hess_use_wide(n);
#define CASE_PROJ(D, R) \
  if (d == D && r == R) { \
    if (wide) { \
      constexpr int PB = 256 / ((D + 1) * R); \
      hipLaunchKernelGGL((k_proj_wide<0, D, R, CF>), \
                         dim3(blocks_for(n, PB)), dim3(256), 0, s, \
                         X, V, G, out, dotWith, ctrl, n, dot_slot, \
                         dot_slot2, guard); \
    } else { \
      hipLaunchKernelGGL((k_proj_dots<0, D, R, CF>), dim3(grid), \
                         dim3(256), 0, s, X, V, G, out, dotWith, ctrl, \
                         n, dot_slot, dot_slot2, guard); \
    } \
    return; \
  }
  DPO_FOREACH_DR(CASE_PROJ)
#undef CASE_PROJ
  dpo_bad_shape(d, r);
}

static void launch_proj_dots(const double* X, const double* V,
                             const double* G, double* out,
                             const double* dotWith, double* ctrl,
                             int n, int d, int r, int dot_slot,
                             int dot_slot2, int guard, hipStream_t s) {
  launch_proj_dots_cf<CF_NONE>(X, V, G, out, dotWith, ctrl, n, d, r,
                               dot_slot, dot_slot2, guard, s);
}

static void launch_polar(const double* A, const double* B, const double* C,
                         double ca, double cb, double cc, double* out,
                         int n, int d, int r, const double* ctrl, int guard,
                         hipStream_t s) {
  const int grid = blocks_for(n, 256);
#define CASE_POLAR(D, R) \
  if (d == D && r == R) { \
    hipLaunchKernelGGL((k_polar_affine<D, R>), dim3(grid), dim3(256), 0, s, \
                       A, B, C, ca, cb, cc, out, n, ctrl, guard); \
    return; \
  }
  DPO_FOREACH_DR(CASE_POLAR)
#undef CASE_POLAR
  dpo_bad_shape(d, r);
}

// Wide-variant selection: element-per-thread (dh*r threads per pose)
// for large agents, where the thread-per-pose kernel's n/64 waves are
// too few to hide HBM latency (1M-pose profile, round 1); the fused
// thread-per-pose kernel stays the default for the (launch-latency-
// bound) small-agent regime. DPO_HESS_WIDE=0/1 forces either path.
static inline bool hess_use_wide(int n) {
  static const int mode = []() {
    const char* v = getenv("DPO_HESS_WIDE");
    return v ? atoi(v) : -1;
  }();
  if (mode == 0) return false;
  if (mode == 1) return true;
  // Measured (profiles/r2c_*): the thread-per-pose kernel runs at
  // ~HBM roofline even at 125k poses/agent (concurrent agents supply
  // the missing waves), while the element-per-thread layout pays ~5x
  // redundant Q-row reads. Wide stays opt-in for experiments.
  return false;
}

template <int MODE, int CF = CF_NONE>
static void launch_hess_fused(const int* rp, const int* ci,
                              const double* vals, const double* V,
                              const double* X, const double* G,
                              double* out, const double* dotW,
                              double* ctrl, int n, int d, int r,
                              int dot_slot, int dot_slot2, int guard,
                              hipStream_t s, int dot_slot3 = -1) {
  const int grid = blocks_for(n, 256);
  const bool wide = hess_use_wide(n);
#define CASE_HF(D, R) \
  if (d == D && r == R) { \
    if (wide) { \
      constexpr int PB = 256 / ((D + 1) * R); \
      hipLaunchKernelGGL((k_hess_wide<D, R, MODE, CF>), \
                         dim3(blocks_for(n, PB)), dim3(256), 0, s, \
                         rp, ci, vals, V, X, G, out, dotW, \
                         ctrl, n, dot_slot, dot_slot2, dot_slot3, \
                         guard); \
    } else { \
      hipLaunchKernelGGL((k_hess_fused<D, R, MODE, CF>), dim3(grid), \
                         dim3(256), 0, s, rp, ci, vals, V, X, G, out, \
                         dotW, ctrl, n, dot_slot, dot_slot2, dot_slot3, \
                         guard); \
    } \
    return; \
  }
  DPO_FOREACH_DR(CASE_HF)
#undef CASE_HF
  dpo_bad_shape(d, r);
}

// Graph-safe zero / device->pinned-host copy. hipMemsetAsync and
// hipMemcpyAsync become SDMA-engine nodes inside a captured hipGraph;
// compute<->SDMA dependencies in replayed graphs are part of the same
// ROCm 7.2 ordering fragility the launch fences work around, so the
// captured bodies use plain kernels only.
__global__ void k_dzero(double* p, long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] = 0.0;
}
__global__ void k_ctrl_to_host(const double* __restrict__ src,
                               double* __restrict__ dst, int n) {
  int i = threadIdx.x;
  if (i < n) dst[i] = src[i];
}

static void launch_precond_dense(const float* Minv, const double* V,
                                 double* Z, int N, int r,
                                 const double* ctrl, int guard,
                                 hipStream_t s) {
  const int gx = blocks_for(N, 256);
  // j-split so small problems still produce >= ~512 workgroups
  int jsplit = 1;
  while (gx * jsplit < 512 && jsplit < 32) jsplit *= 2;
  if (jsplit > 1) {
    // kernel (not hipMemsetAsync): this runs inside captured graphs,
    // which must stay free of SDMA nodes (see fence comment)
    hipLaunchKernelGGL(k_dzero, dim3(blocks_for((long)N * r, 256)),
                       dim3(256), 0, s, Z, (long)N * r);
  }
#define CASE_PD(D, R) \
  if (r == R) { \
    hipLaunchKernelGGL((k_precond_dense<R>), dim3(gx, jsplit), dim3(256), \
                       0, s, Minv, V, Z, N, ctrl, guard); \
    return; \
  }
  DPO_FOREACH_DR(CASE_PD)
#undef CASE_PD
  dpo_bad_shape(-1, r);
}

static void launch_precond_jacobi(const double* L, const double* V,
                                  double* Z, int n, int dh, int r,
                                  const double* ctrl, int guard,
                                  hipStream_t s) {
  const int grid = blocks_for((long)n * r, 256);
  if (dh == 3)
    hipLaunchKernelGGL((k_precond_jacobi<3>), dim3(grid), dim3(256), 0, s,
                       L, V, Z, n, r, ctrl, guard);
  else if (dh == 4)
    hipLaunchKernelGGL((k_precond_jacobi<4>), dim3(grid), dim3(256), 0, s,
                       L, V, Z, n, r, ctrl, guard);
  else
    dpo_bad_shape(dh - 1, r);
}

extern "C" {

void dpo_bsr_spmm(const int* row_ptr, const int* col_idx, const double* vals,
                  int n, int dh, const double* X, double* out, int r,
                  const double* ctrl, int guard, void* stream) {
  launch_spmm(row_ptr, col_idx, vals, n, dh - 1, r, X, out, ctrl, guard,
              (hipStream_t)stream);
}

void dpo_bsr_spmm_mfma(const int* grp_ptr, const int* grp_cols,
                       const int* grp_blk, const double* vals,
                       const double* X, double* out, int ngroups, int n,
                       int r, void* stream) {
  const int waves_per_block = 4;
  const int grid = (ngroups + waves_per_block - 1) / waves_per_block;
  hipLaunchKernelGGL(k_bsr_spmm_mfma_d3, dim3(grid),
                     dim3(64 * waves_per_block), 0, (hipStream_t)stream,
                     grp_ptr, grp_cols, grp_blk, vals, X, out, ngroups,
                     n, r);
}

void dpo_proj_dots(const double* X, const double* V, const double* G,
                   double* out, const double* dotWith, double* ctrl,
                   int n, int d, int r, int dot_slot, int dot_slot2,
                   int neg, int guard, void* stream) {
  (void)neg;  // the negated variant is realized by k_tcg_delta instead
  launch_proj_dots(X, V, G, out, dotWith, ctrl, n, d, r, dot_slot,
                   dot_slot2, guard, (hipStream_t)stream);
}

void dpo_polar_affine(const double* A, const double* B, const double* C,
                      double ca, double cb, double cc, double* out,
                      int n, int d, int r, const double* ctrl, int guard,
                      void* stream) {
  launch_polar(A, B, C, ca, cb, cc, out, n, d, r, ctrl, guard,
               (hipStream_t)stream);
}

void dpo_precond_dense(const float* Minv, const double* V, double* Z,
                       int N, int r, const double* ctrl, int guard,
                       void* stream) {
  launch_precond_dense(Minv, V, Z, N, r, ctrl, guard, (hipStream_t)stream);
}

void dpo_precond_jacobi(const double* L, const double* V, double* Z,
                        int n, int dh, int r, const double* ctrl, int guard,
                        void* stream) {
  launch_precond_jacobi(L, V, Z, n, dh, r, ctrl, guard,
                        (hipStream_t)stream);
}

void dpo_tcg_update(double* eta, double* rvec, const double* delta,
                    const double* Hd, double* eta_snap, double* delta_snap,
                    double* ctrl, long total, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL((k_tcg_update<CF_NONE>),
                     dim3(blocks_for((total + 3) / 4, 256)),
                     dim3(256), 0, s, eta, rvec, delta, Hd, eta_snap,
                     delta_snap, ctrl, total);
}

void dpo_tcg_delta(double* delta, const double* z, double* ctrl, long total,
                   void* stream) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(k_tcg_delta, dim3(blocks_for(total, 256)), dim3(256),
                     0, s, delta, z, ctrl, total);
}

void dpo_form_step(double* step, const double* eta, const double* eta_snap,
                   const double* delta_snap, const double* ctrl, long total,
                   void* stream) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(k_form_step, dim3(blocks_for(total, 256)), dim3(256),
                     0, s, step, eta, eta_snap, delta_snap, ctrl, total);
}

void dpo_axpby(const double* A, const double* B, double a, double b,
               double* out, long total, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(k_axpby, dim3(blocks_for(total, 256)), dim3(256),
                     0, s, A, B, a, b, out, total);
}

void dpo_dots(const double* A, const double* B, const double* C,
              double* ctrl, int slot, int slot2, long total, int guard,
              void* stream) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL((k_dots<CF_NONE>), dim3(blocks_for(total, 256)),
                     dim3(256), 0, s, A, B, C, ctrl, slot, slot2, total,
                     guard, (double*)nullptr);
}

void dpo_ctrl_init(double* ctrl, double tol, double Delta0, double theta,
                   double kappa, void* stream) {
  hipLaunchKernelGGL(k_ctrl_init, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, ctrl, tol, Delta0, theta, kappa);
}
void dpo_ctrl_z0(double* ctrl, void* stream) {
  hipLaunchKernelGGL(k_ctrl_z0, dim3(1), dim3(64), 0, (hipStream_t)stream,
                     ctrl);
}
void dpo_ctrl_alpha(double* ctrl, void* stream) {
  hipLaunchKernelGGL(k_ctrl_alpha, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, ctrl);
}
void dpo_ctrl_rr(double* ctrl, void* stream) {
  hipLaunchKernelGGL(k_ctrl_rr, dim3(1), dim3(64), 0, (hipStream_t)stream,
                     ctrl);
}
void dpo_ctrl_beta(double* ctrl, void* stream) {
  hipLaunchKernelGGL(k_ctrl_beta, dim3(1), dim3(64), 0, (hipStream_t)stream,
                     ctrl);
}
void dpo_ctrl_tcg_end(double* ctrl, void* stream) {
  hipLaunchKernelGGL(k_ctrl_tcg_end, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, ctrl);
}
void dpo_ctrl_candidate(double* ctrl, void* stream) {
  hipLaunchKernelGGL(k_ctrl_candidate, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, ctrl);
}
void dpo_ctrl_accept(double* ctrl, double accept_rho, void* stream) {
  hipLaunchKernelGGL(k_ctrl_accept, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, ctrl, accept_rho);
}
void dpo_ctrl_shrink(double* ctrl, void* stream) {
  hipLaunchKernelGGL(k_ctrl_shrink, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, ctrl);
}

void dpo_q_assemble(double* vals, const double* blocks, const long* slots,
                    const long* edge_of, const double* w, int ncontrib,
                    int bsz, long nnzb, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  DPO_CHECK(hipMemsetAsync(vals, 0, (size_t)nnzb * bsz * sizeof(double), s));
  hipLaunchKernelGGL(k_q_assemble,
                     dim3(blocks_for((long)ncontrib * bsz, 256)), dim3(256),
                     0, s, vals, blocks, slots, edge_of, w, ncontrib, bsz);
}

void dpo_g_assemble(double* Gt, const double* E0, const long* local_pose,
                    const long* nbr_slot, const double* nbr, const double* w,
                    int ne, int dh, int r, long N, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  DPO_CHECK(hipMemsetAsync(Gt, 0, (size_t)N * r * sizeof(double), s));
  hipLaunchKernelGGL(k_g_assemble,
                     dim3(blocks_for((long)ne * dh * r, 256)), dim3(256),
                     0, s, Gt, E0, local_pose, nbr_slot, nbr, w, ne, dh, r);
}

int dpo_ctrl_size() { return CTRL_SIZE; }

void dpo_gnc_weights(const double* X, const double* nbr, const long* e1_idx,
                     const unsigned char* e1_nbr, const long* e2_idx,
                     const unsigned char* e2_nbr, const double* Rm,
                     const double* tm, const double* kappa,
                     const double* tau, const unsigned char* update_mask,
                     const long* widx, double* weights, int ne, int d,
                     int r, double mu, double barc_sq, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  const int grid = blocks_for(ne, 256);
#define CASE_GNC(D, R) \
  if (d == D && r == R) { \
    hipLaunchKernelGGL((k_gnc_weights<D, R>), dim3(grid), dim3(256), 0, s, \
                       X, nbr, e1_idx, e1_nbr, e2_idx, e2_nbr, Rm, tm, \
                       kappa, tau, update_mask, widx, weights, ne, mu, \
                       barc_sq); \
    return; \
  }
  DPO_FOREACH_DR(CASE_GNC)
#undef CASE_GNC
  dpo_bad_shape(d, r);
}

}  // extern "C"

// =====================================================================
// C++ solver orchestration: the full RBCD trust-region local solve and
// the per-round evaluation, enqueued from native code (one Python call
// per solve instead of ~100). This is the MI355X-native runtime
// replacement for the reference's ROPTLIB-driven QuadraticOptimizer
// (QuadraticOptimizer.cpp:61-122).
// =====================================================================

// combine kernel: out[0..2] = [f, 0.5<X,G>, gn2] from dot slots
__global__ void k_eval_combine(double* __restrict__ ctrl,
                               double* __restrict__ out) {
  if (threadIdx.x != 0) return;
  // C_DOT0 = <QX + G, X>, C_DOT2 = <G, X>, C_DOT1 = ||P_X(QX+G)||^2
  out[0] = 0.5 * (ctrl[C_DOT0] + ctrl[C_DOT2]);   // f(X)
  out[1] = 0.5 * ctrl[C_DOT2];                    // 0.5 <X, G>
  out[2] = ctrl[C_DOT1];                          // gradnorm^2
  // self-clean the dot slots for the next eval: saves the standalone
  // 32-byte dzero launch per evaluation (12.8% of bench-scale kernel
  // launches were k_dzero calls, profiles/bench_r2_kernel_stats.csv).
  // Solves re-zero the whole control block themselves; the ctx
  // allocation is zero-initialized, so the invariant "dot slots are
  // zero on eval entry" holds from the start.
  ctrl[C_DOT0] = 0.0;
  ctrl[C_DOT1] = 0.0;
  ctrl[C_DOT2] = 0.0;
  ctrl[C_DOT3] = 0.0;
}

struct DpoCtx {
  int n, d, r, dh, N, max_inner;
  long total;
  double *W, *grad, *eta, *delta, *rvec, *z, *Hd, *step, *Xprop;
  double *eta_snap, *delta_snap, *ctrl;
  double *ctrl_host;      // pinned
  double *ctrl_host_dev;  // device-side view of ctrl_host (mapped)
  // borrowed problem pointers (owned by torch tensors on the Python side)
  const int *q_rp = nullptr, *q_ci = nullptr;
  const double *q_vals = nullptr;
  const double *Gt = nullptr;
  const float *Minv = nullptr;
  const double *Ljac = nullptr;
  // G-assembly structure (owned by torch tensors)
  const double *g_E0 = nullptr;
  const long *g_local_pose = nullptr, *g_nbr_slot = nullptr;
  const double *g_w = nullptr;
  int g_ne = 0;
  double *G_buf = nullptr;  // ctx-owned (N, r) linear-term buffer
  // hipGraph caches for the fixed-pointer round sequences. Keyed by the
  // operand pointers; invalidated whenever set_problem/set_gdata change
  // them (preconditioner refresh, Q rebuild keep the same buffers).
  hipGraphExec_t solve_graph = nullptr;
  const void* solve_key[4] = {};
  int solve_replays = 0;
  hipGraphExec_t eval_graph = nullptr;
  const void* eval_key[4] = {};
  int eval_replays = 0;
  // Periodic graph-exec refresh. The "degradation after a few hundred
  // replays" this originally worked around is now attributed to the
  // SDMA-node / launch-ordering bugs fixed by the fences and
  // kernels-only bodies above; the refresh is kept as cheap insurance
  // (amortized ~1%) and is tunable via DPO_MAX_REPLAYS.
  static int max_replays() {
    static const int v = [] {
      const char* e = getenv("DPO_MAX_REPLAYS");
      return e ? atoi(e) : 128;
    }();
    return v;
  }
  // private stream used only for RECORDING captures (the legacy default
  // stream cannot be captured); graphs replay on the caller's stream.
  hipStream_t cap_stream = nullptr;
  // per-agent execution stream + events: lets concurrent agents' solves
  // and the per-round evaluations overlap on the GPU (colored schedule,
  // 8-agent eval fan-out) while staying ordered against the caller's
  // (torch) stream via events.
  hipStream_t exec_stream = nullptr;
  hipEvent_t start_event = nullptr;
  hipEvent_t done_event = nullptr;
  // state of an in-flight async solve
  double pend_tol = 0, pend_Delta0 = 0, pend_rho = 0;
  double* pend_X = nullptr;
  // data-flow fence slots (see k_fence_* above)
  unsigned int* fences = nullptr;
  // grid-barrier state for the persistent tCG kernel: [count, generation]
  unsigned int* gbar = nullptr;
  // fixed 3-double eval destination for the group fan-out path (a
  // stable pointer keeps the eval graph cache hot)
  double* eval3 = nullptr;
  void invalidate_graphs() {
    if (solve_graph) {
      dpo_ignore(hipGraphExecDestroy(solve_graph));
      solve_graph = nullptr;
    }
    if (eval_graph) {
      dpo_ignore(hipGraphExecDestroy(eval_graph));
      eval_graph = nullptr;
    }
  }
};

// --- data-flow fences around hipGraph launches ----------------------
// ROCm 7.2 on gfx950: a hipGraphLaunch intermittently fails to order
// against work enqueued on other streams via hipStreamWaitEvent (and,
// on the legacy stream, against prior eager work), executing the graph
// early/concurrently. Observed as rare stale/garbage reads in the
// round pipeline (bisected: eager-everything is stable, every failing
// configuration launches a graph whose cross-stream ordering is
// load-bearing). Instead of depending on launch ordering at all, each
// graph carries explicit data-flow fences:
//   producer stream:  k_fence_signal(IN)   (eager kernel — reliable)
//   graph first node: k_fence_wait(IN)     (spins until signalled)
//   graph last node:  k_fence_signal(OUT)
//   consumer stream:  k_fence_wait(OUT)    (eager kernel after launch)
// so even a misordered launch blocks on its first node until its
// inputs are ready, and downstream eager work blocks until the graph
// really finished. Cost: ~2 one-wave kernels per launch (<2 us).
enum { F_SOLVE_IN = 0, F_SOLVE_OUT = 1, F_EVAL_IN = 2, F_EVAL_OUT = 3,
       F_NUM = 4 };

__global__ void k_fence_signal(unsigned int* f) {
  if (threadIdx.x == 0)
    __hip_atomic_store(f, 1u, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
}

__global__ void k_fence_wait(unsigned int* f) {
  if (threadIdx.x == 0) {
    long spins = 0;
    while (__hip_atomic_load(f, __ATOMIC_ACQUIRE,
                             __HIP_MEMORY_SCOPE_AGENT) == 0u) {
      __builtin_amdgcn_s_sleep(64);
      // ~50 s at ~1.7us/iter: a lost signal is a programming error —
      // fail loudly (aborted kernel surfaces on the next sync) rather
      // than silently proceeding on unordered data.
      if (++spins > 30000000L) __builtin_trap();
    }
    __hip_atomic_store(f, 0u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  }
}

static inline void fence_signal(DpoCtx* c, int which, hipStream_t s) {
  hipLaunchKernelGGL(k_fence_signal, dim3(1), dim3(64), 0, s,
                     c->fences + which);
}
static inline void fence_wait(DpoCtx* c, int which, hipStream_t s) {
  hipLaunchKernelGGL(k_fence_wait, dim3(1), dim3(64), 0, s,
                     c->fences + which);
}

static inline void dzero(double* p, long n, hipStream_t s) {
  hipLaunchKernelGGL(k_dzero, dim3((unsigned)((n + 255) / 256)), dim3(256),
                     0, s, p, n);
}

static void ctx_assemble_g(DpoCtx* c, const double* nbr, hipStream_t s) {
  if (c->g_ne == 0) { c->Gt = nullptr; return; }
  dzero(c->G_buf, c->total, s);
  hipLaunchKernelGGL(k_g_assemble,
                     dim3(((long)c->g_ne * c->dh * c->r + 255) / 256),
                     dim3(256), 0, s, c->G_buf, c->g_E0, c->g_local_pose,
                     c->g_nbr_slot, nbr, c->g_w, c->g_ne, c->dh, c->r);
  c->Gt = c->G_buf;
}

static void ctx_precond(DpoCtx* c, const double* V, double* Z,
                        hipStream_t s) {
  if (c->Minv)
    launch_precond_dense(c->Minv, V, Z, c->N, c->r, c->ctrl, ST_RUN, s);
  else
    launch_precond_jacobi(c->Ljac, V, Z, c->n, c->dh, c->r, c->ctrl,
                          ST_RUN, s);
}

// Fused z-stage: z = P_X(M^-1 r) + <z, r> dot + CF tail in one kernel
// when the block-Jacobi preconditioner is active (large agents);
// otherwise the dense apply runs standalone followed by the fused
// projection kernel, as before.
template <int CF>
static void ctx_precond_proj(DpoCtx* c, const double* X, hipStream_t s) {
  if (c->Minv == nullptr) {
    const int grid = blocks_for(c->n, 256);
#define CASE_PP(D, R) \
    if (c->d == D && c->r == R) { \
      hipLaunchKernelGGL((k_precond_proj<D, R, CF>), dim3(grid), \
                         dim3(256), 0, s, c->Ljac, c->rvec, X, c->z, \
                         c->ctrl, c->n, C_DOT0, ST_RUN); \
      return; \
    }
    DPO_FOREACH_DR(CASE_PP)
#undef CASE_PP
  }
  ctx_precond(c, c->rvec, c->z, s);
  launch_proj_dots_cf<CF>(X, c->z, nullptr, c->z, c->rvec, c->ctrl,
                          c->n, c->d, c->r, C_DOT0, -1, ST_RUN, s);
}

static void ctx_spmm(DpoCtx* c, const double* X, double* out, int guard,
                     hipStream_t s) {
  launch_spmm(c->q_rp, c->q_ci, c->q_vals, c->n, c->d, c->r, X, out,
              c->ctrl, guard, s);
}

extern "C" {

void* dpo_ctx_create(int n, int d, int r, int max_inner) {
  DpoCtx* c = new DpoCtx();
  c->n = n; c->d = d; c->r = r; c->dh = d + 1;
  c->N = c->dh * n; c->max_inner = max_inner;
  c->total = (long)c->N * r;
  size_t vb = (size_t)c->total * sizeof(double);
  DPO_CHECK(hipMalloc(&c->W, vb));
  DPO_CHECK(hipMalloc(&c->grad, vb));
  DPO_CHECK(hipMalloc(&c->eta, vb));
  DPO_CHECK(hipMalloc(&c->delta, vb));
  DPO_CHECK(hipMalloc(&c->rvec, vb));
  DPO_CHECK(hipMalloc(&c->z, vb));
  DPO_CHECK(hipMalloc(&c->Hd, vb));
  DPO_CHECK(hipMalloc(&c->step, vb));
  DPO_CHECK(hipMalloc(&c->Xprop, vb));
  DPO_CHECK(hipMalloc(&c->eta_snap, vb * (max_inner + 1)));
  DPO_CHECK(hipMalloc(&c->delta_snap, vb * (max_inner + 1)));
  DPO_CHECK(hipMalloc(&c->ctrl, CTRL_SIZE * sizeof(double)));
  DPO_CHECK(hipMalloc(&c->G_buf, vb));
  // delta is consumed as `beta*delta - z` with beta == 0 on the first
  // tCG iteration: one-time zeroing keeps 0*garbage (possible NaN bit
  // patterns in fresh allocations) out of the recurrence, and the
  // per-solve zeroing passes are dropped from the captured body
  DPO_CHECK(hipMemset(c->delta, 0, vb));
  DPO_CHECK(hipMemset(c->eta, 0, vb));
  // +16 slots: persistent-kernel stage timestamps (DPO_DBG_PERSIST=1)
  DPO_CHECK(hipHostMalloc(&c->ctrl_host, (CTRL_SIZE + 16) * sizeof(double)));
  DPO_CHECK(hipHostGetDevicePointer((void**)&c->ctrl_host_dev,
                                    c->ctrl_host, 0));
  DPO_CHECK(hipStreamCreateWithFlags(&c->cap_stream,
                                     hipStreamNonBlocking));
  DPO_CHECK(hipStreamCreateWithFlags(&c->exec_stream,
                                     hipStreamNonBlocking));
  DPO_CHECK(hipEventCreateWithFlags(&c->start_event,
                                    hipEventDisableTiming));
  DPO_CHECK(hipEventCreateWithFlags(&c->done_event,
                                    hipEventDisableTiming));
  DPO_CHECK(hipMalloc(&c->fences, F_NUM * sizeof(unsigned int)));
  DPO_CHECK(hipMemset(c->fences, 0, F_NUM * sizeof(unsigned int)));
  DPO_CHECK(hipMalloc(&c->eval3, 3 * sizeof(double)));
  DPO_CHECK(hipMalloc(&c->gbar, 2 * sizeof(unsigned int)));
  DPO_CHECK(hipMemset(c->gbar, 0, 2 * sizeof(unsigned int)));
  return c;
}

void dpo_ctx_destroy(void* h) {
  DpoCtx* c = (DpoCtx*)h;
  dpo_ignore(hipFree(c->W)); dpo_ignore(hipFree(c->grad)); dpo_ignore(hipFree(c->eta)); dpo_ignore(hipFree(c->delta));
  dpo_ignore(hipFree(c->rvec)); dpo_ignore(hipFree(c->z)); dpo_ignore(hipFree(c->Hd)); dpo_ignore(hipFree(c->step));
  dpo_ignore(hipFree(c->Xprop)); dpo_ignore(hipFree(c->eta_snap)); dpo_ignore(hipFree(c->delta_snap));
  c->invalidate_graphs();
  if (c->cap_stream) dpo_ignore(hipStreamDestroy(c->cap_stream));
  if (c->exec_stream) dpo_ignore(hipStreamDestroy(c->exec_stream));
  if (c->start_event) dpo_ignore(hipEventDestroy(c->start_event));
  if (c->done_event) dpo_ignore(hipEventDestroy(c->done_event));
  dpo_ignore(hipFree(c->ctrl)); dpo_ignore(hipFree(c->G_buf)); dpo_ignore(hipHostFree(c->ctrl_host));
  dpo_ignore(hipFree(c->fences)); dpo_ignore(hipFree(c->eval3)); dpo_ignore(hipFree(c->gbar));
  delete c;
}

void dpo_ctx_set_problem(void* h, const int* rp, const int* ci,
                         const double* vals, const double* Gt,
                         const float* Minv, const double* Ljac) {
  DpoCtx* c = (DpoCtx*)h;
  if (c->q_vals != vals || c->Minv != Minv || c->Ljac != Ljac
      || c->Gt != Gt || c->q_rp != rp)
    c->invalidate_graphs();
  c->q_rp = rp; c->q_ci = ci; c->q_vals = vals;
  c->Gt = Gt; c->Minv = Minv; c->Ljac = Ljac;
}

static void launch_solve_persist(DpoCtx* c, const double* X,
                                 const double* nbr, double tol,
                                 double Delta0, double accept_rho,
                                 hipStream_t s, int full) {
  const int gvec = (int)((c->total + 255) / 256);
  // replicate ctx_assemble_g's pointer logic (assembly itself runs
  // inside the kernel); c->Gt must be correct for the host-driven
  // shrink-loop replay in solve_postsync
  const double* G;
  double* Gw = nullptr;
  if (nbr) {
    if (c->g_ne) {
      c->Gt = c->G_buf;
      G = c->G_buf;
      Gw = c->G_buf;
    } else {
      c->Gt = nullptr;
      G = nullptr;
    }
  } else {
    G = c->Gt;
  }
  // NOTE: widening the grid for j-split preconditioner stages was
  // measured SLOWER under multi-agent concurrency (4 agents x 200
  // workgroups oversubscribes the 256 CUs and the idle blocks' barrier
  // spins starve real work) — keep the grid at one block per 256
  // elements and rely on the 4-chain unroll for MLP.
  const int jmul = 1;
#define CASE_TP(D, R) \
  if (c->d == D && c->r == R) { \
    hipLaunchKernelGGL((k_solve_persist<D, R>), dim3(gvec * jmul), \
                       dim3(256), 0, \
                       s, c->q_rp, c->q_ci, c->q_vals, X, G, Gw, c->g_E0, \
                       c->g_local_pose, c->g_nbr_slot, nbr, c->g_w, \
                       c->g_ne, c->Minv, c->eta, c->rvec, c->delta, c->z, \
                       c->Hd, c->eta_snap, c->delta_snap, c->step, \
                       c->Xprop, c->ctrl, c->ctrl_host_dev, c->gbar, \
                       c->n, c->N, c->total, c->max_inner, tol, Delta0, \
                       1.0, 0.1, accept_rho, jmul, full); \
    return; \
  }
  DPO_FOREACH_DR(CASE_TP)
#undef CASE_TP
  dpo_bad_shape(c->d, c->r);
}

// Enqueue the fixed pre-sync solve sequence (gradient, tCG, first
// candidate + acceptance test). Capturable as one hipGraph: every
// operand pointer is stable across rounds by construction.
static void enqueue_solve_body(DpoCtx* c, double* X, const double* nbr,
                               double tol, double Delta0,
                               double accept_rho, hipStream_t s) {
  const int n = c->n, d = c->d, r = c->r;
  const long total = c->total;
  const int gvec = (int)((total + 255) / 256);
  // Fused control tails (CF_*): the tCG scalar decisions run in the
  // last-arriving block of the producing reduction kernel instead of a
  // dedicated single-wave kernel, removing 3 of 8 launches per tCG
  // iteration. The ~4 us/launch dispatch latency floor is what bounds
  // these small-agent solves (see profiles/), so launch count is the
  // lever. An earlier attempt at this fusion showed trajectory scatter
  // that was eventually root-caused to an LDS reuse race in
  // block_reduce_atomic plus the hipGraph ordering bugs (both fixed);
  // DPO_NO_CF=1 restores the dedicated control kernels.
  static const bool no_cf = dpo_env_flag("DPO_NO_CF");
  static const bool no_persist = dpo_env_flag("DPO_NO_PERSIST");
  fence_wait(c, F_SOLVE_IN, s);  // first node: block until inputs ready
  // Persistent-kernel scope: by default the tCG LOOP runs as one
  // kernel (k_solve_persist with full=0) between the grad/z0 and
  // candidate launch sequences — measured fastest under multi-agent
  // concurrency. DPO_PERSIST_FULL=1 fuses the ENTIRE pre-sync solve
  // into the kernel (fewer launches but the in-kernel z0 dense-
  // preconditioner apply is latency-bound on the L2-cold Minv pass,
  // which the standalone j-split kernel hides better).
  static const bool persist_full = dpo_env_flag("DPO_PERSIST_FULL");
  // N cap: the in-kernel dense preconditioner apply (one element per
  // thread, 4 load chains) only beats the standalone j-split kernel
  // while Minv stays cache-resident; for larger problems the
  // per-stage launch path's j-split apply sustains far higher
  // bandwidth (measured: city10000/cubicle agents at N ~ 4600-6000
  // were 4-8x slower under the persistent path).
  // Round 2: the persistent kernel is now OPT-IN (DPO_PERSIST=1).
  // After the gradient/eval fusion cut the staged path's launch count,
  // A/B probes (gpurun_out/r2h_probe.log) measured the staged path at
  // ~1.4 ms/round vs ~2.6 ms/round persistent on the 8-agent sphere2500
  // bench: at 25-workgroup grids the in-kernel grid barriers cost
  // 380-850 us per solve loop under multi-agent concurrency.
  static const bool persist_opt_in = dpo_env_flag("DPO_PERSIST");
  const bool persist_ok = persist_opt_in && !no_persist && !no_cf
      && c->Minv != nullptr
      && c->N <= 1500 && (total + 255) / 256 <= 256;
  if (persist_ok && persist_full) {
    launch_solve_persist(c, X, nbr, tol, Delta0, accept_rho, s, 1);
    fence_signal(c, F_SOLVE_OUT, s);
    return;
  }
  if (nbr) ctx_assemble_g(c, nbr, s);
  dzero(c->ctrl, CTRL_SIZE, s);

  // gradient phase: one fused kernel computes Q@X + G, projects at X
  // and accumulates <P,P> / <QX+G,X> (was spmm + proj_dots — the
  // two-kernel split re-read the full iterate from HBM)
  launch_hess_fused<0>(c->q_rp, c->q_ci, c->q_vals, X, X, c->Gt,
                       c->grad, nullptr, c->ctrl, n, d, r,
                       C_DOT1, C_DOT0, -1, s,
                       c->Gt ? C_DOT2 : -1);  // <G,X> folded in
  hipLaunchKernelGGL(k_axpby, dim3(gvec), dim3(256), 0, s,
                     c->grad, (const double*)nullptr, 1.0, 0.0, c->rvec,
                     total);
  hipLaunchKernelGGL(k_ctrl_init, dim3(1), dim3(64), 0, s, c->ctrl, tol,
                     Delta0, 1.0, 0.1);
  // z0
  if (no_cf) {
    ctx_precond(c, c->rvec, c->z, s);
    launch_proj_dots(X, c->z, nullptr, c->z, c->rvec, c->ctrl, n, d, r,
                     C_DOT0, -1, ST_RUN, s);
    hipLaunchKernelGGL(k_ctrl_z0, dim3(1), dim3(64), 0, s, c->ctrl);
  } else {
    ctx_precond_proj<CF_Z0>(c, X, s);
  }
  hipLaunchKernelGGL(k_tcg_delta, dim3(gvec), dim3(256), 0, s,
                     c->delta, c->z, c->ctrl, total);

  if (persist_ok) {
    // the whole tCG loop as one kernel (grid barriers between stages)
    launch_solve_persist(c, X, nbr, tol, Delta0, accept_rho, s, 0);
  } else
  for (int j = 0; j < c->max_inner; ++j) {
    if (no_cf) {
      launch_hess_fused<0>(c->q_rp, c->q_ci, c->q_vals, c->delta, X,
                           nullptr, c->Hd, c->delta, c->ctrl, n, d, r,
                           C_DOT0, -1, ST_RUN, s);
      hipLaunchKernelGGL(k_ctrl_alpha, dim3(1), dim3(64), 0, s, c->ctrl);
      hipLaunchKernelGGL((k_tcg_update<CF_NONE>),
                         dim3(blocks_for((total + 3) / 4, 256)),
                         dim3(256), 0, s, c->eta, c->rvec, c->delta,
                         c->Hd, c->eta_snap, c->delta_snap, c->ctrl,
                         total);
      hipLaunchKernelGGL(k_ctrl_rr, dim3(1), dim3(64), 0, s, c->ctrl);
    } else {
      launch_hess_fused<0, CF_ALPHA>(
          c->q_rp, c->q_ci, c->q_vals, c->delta, X, nullptr, c->Hd,
          c->delta, c->ctrl, n, d, r, C_DOT0, -1, ST_RUN, s);
      hipLaunchKernelGGL((k_tcg_update<CF_RR>),
                         dim3(blocks_for((total + 3) / 4, 256)),
                         dim3(256), 0, s, c->eta, c->rvec, c->delta,
                         c->Hd, c->eta_snap, c->delta_snap, c->ctrl,
                         total);
    }
    if (no_cf) {
      ctx_precond(c, c->rvec, c->z, s);
      launch_proj_dots(X, c->z, nullptr, c->z, c->rvec, c->ctrl, n, d, r,
                       C_DOT0, -1, ST_RUN, s);
      hipLaunchKernelGGL(k_ctrl_beta, dim3(1), dim3(64), 0, s, c->ctrl);
    } else {
      ctx_precond_proj<CF_BETA>(c, X, s);
    }
    hipLaunchKernelGGL(k_tcg_delta, dim3(gvec), dim3(256), 0, s,
                       c->delta, c->z, c->ctrl, total);
  }
  hipLaunchKernelGGL(k_ctrl_tcg_end, dim3(1), dim3(64), 0, s, c->ctrl);
  // first candidate attempt is part of the fixed sequence
  hipLaunchKernelGGL(k_ctrl_candidate, dim3(1), dim3(64), 0, s, c->ctrl);
  hipLaunchKernelGGL(k_form_step, dim3(gvec), dim3(256), 0, s,
                     c->step, c->eta, c->eta_snap, c->delta_snap,
                     c->ctrl, total);
  launch_polar(X, c->step, nullptr, 1.0, 1.0, 0.0, c->Xprop, n, d, r,
               c->ctrl, ST_TCG_STOP, s);
  launch_hess_fused<1>(c->q_rp, c->q_ci, c->q_vals, c->Xprop, nullptr,
                       c->Gt, nullptr, nullptr, c->ctrl, n, d, r,
                       C_DOT0, C_DOT2, ST_TCG_STOP, s);
  hipLaunchKernelGGL(k_ctrl_accept, dim3(1), dim3(64), 0, s, c->ctrl,
                     accept_rho);
  hipLaunchKernelGGL(k_ctrl_to_host, dim3(1), dim3(64), 0, s, c->ctrl,
                     c->ctrl_host_dev, CTRL_SIZE);
  fence_signal(c, F_SOLVE_OUT, s);  // last node: mark sequence complete
}

// Full RBCD local solve in place on X. stats_out (host, >= 8 doubles):
// [status, f_init, gn_init, f_opt, gn_opt, rho, shrink_count, iters]
// nbr != null: assemble G from the packed neighbor buffer first; the
// whole pre-sync sequence is then served from a cached hipGraph.
// Ensure the solve graph for (X, nbr, tol, Delta0) is cached; run the
// pre-sync sequence (via graph replay or eagerly) on stream s WITHOUT
// the final synchronize. Returns true when the sequence was enqueued
// asynchronously (graph path) — eager fallback syncs internally.
static bool solve_presync(DpoCtx* c, double* X, const double* nbr,
                          double tol, double Delta0, double accept_rho,
                          hipStream_t s) {
  static const bool no_graph = dpo_env_flag("DPO_NO_SOLVE_GRAPH");
  if (no_graph) {
    // fully async eager enqueue; callers sync (finish / impl)
    enqueue_solve_body(c, X, nbr, tol, Delta0, accept_rho, s);
    fence_wait(c, F_SOLVE_OUT, s);
    return true;
  }
  const void* key[4] = {X, nbr, (const void*)(intptr_t)(tol * 1e9),
                        (const void*)(intptr_t)Delta0};
  bool key_match = c->solve_graph && memcmp(key, c->solve_key,
                                            sizeof(key)) == 0
                   && c->solve_replays < DpoCtx::max_replays();
  if (!key_match) {
    if (c->solve_graph) {
      dpo_ignore(hipGraphExecDestroy(c->solve_graph));
      c->solve_graph = nullptr;
    }
    c->solve_replays = 0;
    hipGraph_t graph = nullptr;
    hipStream_t cs_ = c->cap_stream;
    hipError_t rc = hipStreamBeginCapture(
        cs_, hipStreamCaptureModeThreadLocal);
    if (rc == hipSuccess) {
      enqueue_solve_body(c, X, nbr, tol, Delta0, accept_rho, cs_);
      rc = hipStreamEndCapture(cs_, &graph);
    }
    hipStreamCaptureStatus capst = hipStreamCaptureStatusNone;
    dpo_ignore(hipStreamIsCapturing(cs_, &capst));
    if (capst != hipStreamCaptureStatusNone) {
      hipGraph_t dead = nullptr;
      dpo_ignore(hipStreamEndCapture(cs_, &dead));
      if (dead) dpo_ignore(hipGraphDestroy(dead));
    }
    if (rc == hipSuccess && graph) {
      rc = hipGraphInstantiate(&c->solve_graph, graph, nullptr, nullptr, 0);
      dpo_ignore(hipGraphDestroy(graph));
    }
    dpo_ignore(hipGetLastError());  // clear sticky capture-related error state
    if (rc != hipSuccess || !c->solve_graph) {
      // capture unavailable: run eagerly on the caller's stream
      c->solve_graph = nullptr;
      enqueue_solve_body(c, X, nbr, tol, Delta0, accept_rho, s);
      fence_wait(c, F_SOLVE_OUT, s);
      DPO_CHECK(hipStreamSynchronize(s));
      return false;
    }
    memcpy(c->solve_key, key, sizeof(key));
  }
  c->solve_replays++;
  DPO_CHECK(hipGraphLaunch(c->solve_graph, s));
  // eager wait pins downstream stream work (and host syncs) to the
  // graph's actual completion even if the launch misordered
  fence_wait(c, F_SOLVE_OUT, s);
  return true;
}

// Post-sync part of the solve: acceptance decision + (rare) shrink
// loop. Assumes the pre-sync sequence has completed on stream s and
// ctrl_host holds the control block. A rejected first candidate
// enqueues the ENTIRE guarded radius-shrink loop in one batch (each
// attempt's kernels no-op via the ctrl guard once a candidate is
// accepted) behind a single stream sync, instead of a host round-trip
// per attempt.
static int solve_postsync(DpoCtx* c, double* X, double accept_rho,
                          int max_shrink, int compute_final_gn,
                          double* stats_out, hipStream_t s) {
  const int n = c->n, d = c->d, r = c->r;
  const long total = c->total;
  const int gvec = (int)((total + 255) / 256);
  int st = (int)c->ctrl_host[C_STATUS];
  if (st == ST_TCG_STOP && max_shrink > 0) {
    for (int attempt = 1; attempt <= max_shrink; ++attempt) {
      hipLaunchKernelGGL(k_ctrl_shrink, dim3(1), dim3(64), 0, s, c->ctrl);
      hipLaunchKernelGGL(k_ctrl_candidate, dim3(1), dim3(64), 0, s,
                         c->ctrl);
      hipLaunchKernelGGL(k_form_step, dim3(gvec), dim3(256), 0, s,
                         c->step, c->eta, c->eta_snap, c->delta_snap,
                         c->ctrl, total);
      launch_polar(X, c->step, nullptr, 1.0, 1.0, 0.0, c->Xprop, n, d, r,
                   c->ctrl, ST_TCG_STOP, s);
      launch_hess_fused<1>(c->q_rp, c->q_ci, c->q_vals, c->Xprop, nullptr,
                           c->Gt, nullptr, nullptr, c->ctrl, n, d, r,
                           C_DOT0, C_DOT2, ST_TCG_STOP, s);
      hipLaunchKernelGGL(k_ctrl_accept, dim3(1), dim3(64), 0, s, c->ctrl,
                         accept_rho);
    }
    hipLaunchKernelGGL(k_ctrl_to_host, dim3(1), dim3(64), 0, s, c->ctrl,
                       c->ctrl_host_dev, CTRL_SIZE);
    DPO_CHECK(hipStreamSynchronize(s));
    st = (int)c->ctrl_host[C_STATUS];
  }
  const int shrinks = (int)c->ctrl_host[C_SHRINKS];
  int status = st;
  if (st == ST_ACCEPTED) {
    DPO_CHECK(hipMemcpyAsync(X, c->Xprop, total * sizeof(double),
                             hipMemcpyDeviceToDevice, s));
  } else if (st == ST_TCG_STOP) {
    status = ST_GIVE_UP;  // every radius rejected: keep the iterate
  }

  double f_init = c->ctrl_host[C_FX];
  double gn_init = sqrt(c->ctrl_host[C_GN0SQ]);
  double f_opt = (status == ST_ACCEPTED) ? c->ctrl_host[C_FPROP] : f_init;
  double gn_opt = gn_init;
  if (compute_final_gn && status == ST_ACCEPTED) {
    DPO_CHECK(hipMemsetAsync(c->ctrl + C_DOT1, 0, sizeof(double), s));
    launch_hess_fused<0>(c->q_rp, c->q_ci, c->q_vals, X, X, c->Gt,
                         c->grad, nullptr, c->ctrl, n, d, r, C_DOT1, -1,
                         -1, s);
    DPO_CHECK(hipMemcpyAsync(c->ctrl_host + C_DOT1, c->ctrl + C_DOT1,
                             sizeof(double), hipMemcpyDeviceToHost, s));
    DPO_CHECK(hipStreamSynchronize(s));
    gn_opt = sqrt(c->ctrl_host[C_DOT1]);
  }
  if (stats_out) {
    stats_out[0] = status;
    stats_out[1] = f_init;
    stats_out[2] = gn_init;
    stats_out[3] = f_opt;
    stats_out[4] = gn_opt;
    stats_out[5] = c->ctrl_host[C_RHO];
    stats_out[6] = shrinks;
    stats_out[7] = c->ctrl_host[C_HLEN];
  }
  return status;
}

static int rbcd_solve_impl(DpoCtx* c, double* X, const double* nbr,
                           double tol, double Delta0, int max_shrink,
                           double accept_rho, int compute_final_gn,
                           double* stats_out, hipStream_t s) {
  fence_signal(c, F_SOLVE_IN, s);
  bool async = solve_presync(c, X, nbr, tol, Delta0, accept_rho, s);
  if (async) DPO_CHECK(hipStreamSynchronize(s));
  return solve_postsync(c, X, accept_rho, max_shrink, compute_final_gn,
                        stats_out, s);
}

int dpo_rbcd_solve(void* h, double* X, double tol, double Delta0,
                   int max_shrink, double accept_rho,
                   int compute_final_gn, double* stats_out, void* stream) {
  return rbcd_solve_impl((DpoCtx*)h, X, nullptr, tol, Delta0, max_shrink,
                         accept_rho, compute_final_gn, stats_out,
                         (hipStream_t)stream);
}

void dpo_round_eval(void* h, const double* X, const double* nbr,
                    double* out_dev, void* stream);  // fwd decl
static void round_eval_impl(DpoCtx* c, const double* X, const double* nbr,
                            double* out_dev, hipStream_t s,
                            bool wait_out = true);  // fwd decl

// --- async (multi-stream) round entry points ------------------------
// Launch the solve's pre-sync sequence on the ctx's private execution
// stream, ordered after the caller's stream. Finish with
// dpo_round_solve_finish. Concurrent agents overlap on the GPU.
void dpo_round_solve_async(void* h, double* X, const double* nbr,
                           double tol, double Delta0, double accept_rho,
                           void* join_stream) {
  DpoCtx* c = (DpoCtx*)h;
  hipStream_t js = (hipStream_t)join_stream;
  // the IN fence is signalled on the PRODUCER (torch) stream: it
  // carries the completion of the boundary-pose scatter into the solve
  // sequence as a data dependency (see fence comment above). No event
  // ceremony — the fence IS the ordering mechanism (launch-order
  // hipStreamWaitEvent is exactly what graph launches fail to honor).
  fence_signal(c, F_SOLVE_IN, js);
  solve_presync(c, X, nbr, tol, Delta0, accept_rho, c->exec_stream);
  c->pend_X = X;
  c->pend_tol = tol;
  c->pend_Delta0 = Delta0;
  c->pend_rho = accept_rho;
}

int dpo_round_solve_finish(void* h, int max_shrink, double* stats_out,
                           void* join_stream) {
  DpoCtx* c = (DpoCtx*)h;
  DPO_CHECK(hipStreamSynchronize(c->exec_stream));
  static const bool dbg_persist = dpo_env_flag("DPO_DBG_PERSIST");
  if (dbg_persist) {
    const double* ts = c->ctrl_host + CTRL_SIZE;
    fprintf(stderr, "[persist us] zeroG %.1f grad %.1f z0 %.1f loop %.1f"
            " end %.1f cand %.1f pub %.1f\n",
            (ts[1]-ts[0])/100.0, (ts[2]-ts[1])/100.0, (ts[3]-ts[2])/100.0,
            (ts[4]-ts[3])/100.0, (ts[5]-ts[4])/100.0, (ts[6]-ts[5])/100.0,
            (ts[7]-ts[6])/100.0);
  }
  int st = solve_postsync(c, c->pend_X, c->pend_rho, max_shrink, 0,
                          stats_out, c->exec_stream);
  DPO_CHECK(hipEventRecord(c->done_event, c->exec_stream));
  DPO_CHECK(hipStreamWaitEvent((hipStream_t)join_stream, c->done_event, 0));
  return st;
}

void dpo_round_eval_async(void* h, const double* X, const double* nbr,
                          double* out_dev, void* join_stream) {
  DpoCtx* c = (DpoCtx*)h;
  hipStream_t js = (hipStream_t)join_stream;
  // IN fence on the producer stream: carries the boundary-pose scatter
  fence_signal(c, F_EVAL_IN, js);
  round_eval_impl(c, X, nbr, out_dev, c->exec_stream, true);
  DPO_CHECK(hipEventRecord(c->done_event, c->exec_stream));
}

void dpo_eval_join(void* h, void* join_stream) {
  DpoCtx* c = (DpoCtx*)h;
  DPO_CHECK(hipStreamWaitEvent((hipStream_t)join_stream, c->done_event, 0));
}

// Per-round evaluation: out_dev (>=3 doubles, device) = [f, 0.5<X,G>, gn2]
// using the ctx's problem pointers. No host sync.
void dpo_eval_terms(void* h, const double* X, double* out_dev,
                    void* stream) {
  DpoCtx* c = (DpoCtx*)h;
  hipStream_t s = (hipStream_t)stream;
  const int n = c->n, d = c->d, r = c->r;
  const long total = c->total;
  const int gvec = (int)((total + 255) / 256);
  static const bool no_cf = dpo_env_flag("DPO_NO_CF");
  (void)gvec; (void)total; (void)no_cf;
  // dot slots are pre-zeroed (zero-init at allocation; k_eval_combine
  // self-cleans after each eval; solves wipe the whole control block)
  // ONE fused kernel: Q@X + G, projection at X, and all three scalars
  // <P,P> / <QX+G,X> / <G,X> (the standalone <G,X> k_dots re-read
  // 2x the iterate from HBM and was the eval phase's fattest kernel
  // at 1M poses — profiles/r2c_narrow)
  launch_hess_fused<0>(c->q_rp, c->q_ci, c->q_vals, X, X, c->Gt,
                       c->grad, nullptr, c->ctrl, n, d, r,
                       C_DOT1, C_DOT0, -1, s, c->Gt ? C_DOT2 : -1);
  hipLaunchKernelGGL(k_eval_combine, dim3(1), dim3(64), 0, s, c->ctrl,
                     out_dev);
}


// Install the static G-assembly structure (per-agent, set once; weights
// pointer re-read every assembly so GNC re-weighting is free).
void dpo_ctx_set_gdata(void* h, const double* E0, const long* local_pose,
                       const long* nbr_slot, const double* w, int ne) {
  DpoCtx* c = (DpoCtx*)h;
  if (c->g_E0 != E0 || c->g_w != w) c->invalidate_graphs();
  c->g_E0 = E0; c->g_local_pose = local_pose; c->g_nbr_slot = nbr_slot;
  c->g_w = w; c->g_ne = ne;
}

// Fused per-round entry points: assemble G from the packed neighbor
// buffer, then solve / evaluate. One Python call each.
int dpo_round_solve(void* h, double* X, const double* nbr, double tol,
                    double Delta0, int max_shrink, double accept_rho,
                    double* stats_out, void* stream) {
  return rbcd_solve_impl((DpoCtx*)h, X, nbr, tol, Delta0, max_shrink,
                         accept_rho, 0, stats_out, (hipStream_t)stream);
}

// Non-caching eval enqueue (for embedding in an externally captured
// graph, e.g. one driver-level graph covering every agent's eval).
// Deliberately unfenced: the embedding graph owns the ordering.
void dpo_round_eval_raw(void* h, const double* X, const double* nbr,
                        double* out_dev, void* stream) {
  DpoCtx* c = (DpoCtx*)h;
  ctx_assemble_g(c, nbr, (hipStream_t)stream);
  dpo_eval_terms(h, X, out_dev, stream);
}

// Fenced eval body: wait on the IN fence, assemble G + evaluate, signal
// the OUT fence. Used for both capture and eager enqueue.
static void enqueue_eval_body(DpoCtx* c, const double* X, const double* nbr,
                              double* out_dev, hipStream_t s) {
  fence_wait(c, F_EVAL_IN, s);
  ctx_assemble_g(c, nbr, s);
  dpo_eval_terms((void*)c, X, out_dev, (void*)s);
  fence_signal(c, F_EVAL_OUT, s);
}

// IN fence must already be signalled on the producer stream. With
// wait_out=false the caller consumes the OUT fence itself (group
// fan-out path: one gather kernel waits on every agent's OUT fence).
static void round_eval_impl(DpoCtx* c, const double* X, const double* nbr,
                            double* out_dev, hipStream_t s,
                            bool wait_out) {
  static const bool no_graph = dpo_env_flag("DPO_NO_EVAL_GRAPH");
  if (no_graph) {
    enqueue_eval_body(c, X, nbr, out_dev, s);
    if (wait_out) fence_wait(c, F_EVAL_OUT, s);
    return;
  }
  const void* key[4] = {X, nbr, out_dev, nullptr};
  bool match = c->eval_graph && memcmp(key, c->eval_key, sizeof(key)) == 0
               && c->eval_replays < DpoCtx::max_replays();
  if (!match) {
    if (c->eval_graph) {
      dpo_ignore(hipGraphExecDestroy(c->eval_graph));
      c->eval_graph = nullptr;
    }
    c->eval_replays = 0;
    hipGraph_t graph = nullptr;
    hipStream_t cs_ = c->cap_stream;
    hipError_t rc = hipStreamBeginCapture(
        cs_, hipStreamCaptureModeThreadLocal);
    if (rc == hipSuccess) {
      enqueue_eval_body(c, X, nbr, out_dev, cs_);
      rc = hipStreamEndCapture(cs_, &graph);
    }
    hipStreamCaptureStatus capst = hipStreamCaptureStatusNone;
    dpo_ignore(hipStreamIsCapturing(cs_, &capst));
    if (capst != hipStreamCaptureStatusNone) {
      hipGraph_t dead = nullptr;
      dpo_ignore(hipStreamEndCapture(cs_, &dead));
      if (dead) dpo_ignore(hipGraphDestroy(dead));
    }
    if (rc == hipSuccess && graph) {
      rc = hipGraphInstantiate(&c->eval_graph, graph, nullptr, nullptr, 0);
      dpo_ignore(hipGraphDestroy(graph));
    }
    dpo_ignore(hipGetLastError());  // clear sticky capture-related error state
    if (rc != hipSuccess || !c->eval_graph) {
      c->eval_graph = nullptr;
      enqueue_eval_body(c, X, nbr, out_dev, s);
      if (wait_out) fence_wait(c, F_EVAL_OUT, s);
      return;
    }
    memcpy(c->eval_key, key, sizeof(key));
  }
  c->eval_replays++;
  DPO_CHECK(hipGraphLaunch(c->eval_graph, s));
  if (wait_out) fence_wait(c, F_EVAL_OUT, s);
}

void dpo_round_eval(void* h, const double* X, const double* nbr,
                    double* out_dev, void* stream) {
  DpoCtx* c = (DpoCtx*)h;
  hipStream_t s = (hipStream_t)stream;
  fence_signal(c, F_EVAL_IN, s);
  round_eval_impl(c, X, nbr, out_dev, s);
}

// --- multi-agent round fan-out --------------------------------------
// The distributed driver runs one rank per GPU; within a rank several
// agents share the device. A round's per-agent enqueue loops (solve
// fan-out over the active color, eval fan-out over every agent) are
// pure launch overhead when driven from Python (~35 us/agent of
// interpreter + dispatcher time); DpoGroup runs each loop as ONE C
// call and gathers the per-agent eval scalars with a single kernel.

__global__ void k_gather3(double* __restrict__ out, long stride,
                          double* const* __restrict__ srcs,
                          const int* __restrict__ rows, int n) {
  int i = blockIdx.x;
  int j = threadIdx.x;
  if (i < n && j < 3) out[(long)rows[i] * stride + j] = srcs[i][j];
}

// Signal every agent's eval IN fence with one launch (lane per agent).
__global__ void k_fence_signal_many(unsigned int* const* __restrict__ f,
                                    int n) {
  int i = threadIdx.x;
  if (i < n)
    __hip_atomic_store(f[i], 1u, __ATOMIC_RELEASE,
                       __HIP_MEMORY_SCOPE_AGENT);
}

// Wait on & consume every agent's eval OUT fence, then gather each
// agent's 3 eval scalars (agent-scope loads: the producers ran on
// other XCDs) — replaces n (event-record + event-wait + copy) chains.
__global__ void k_gather3_fenced(double* __restrict__ out, long stride,
                                 double* const* __restrict__ srcs,
                                 const int* __restrict__ rows,
                                 unsigned int* const* __restrict__ f,
                                 int n) {
  int i = threadIdx.x;
  if (i < n) {
    long spins = 0;
    while (__hip_atomic_load(f[i], __ATOMIC_ACQUIRE,
                             __HIP_MEMORY_SCOPE_AGENT) == 0u) {
      __builtin_amdgcn_s_sleep(32);
      if (++spins > 30000000L) __builtin_trap();
    }
    __hip_atomic_store(f[i], 0u, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_AGENT);
    #pragma unroll
    for (int j = 0; j < 3; ++j)
      out[(long)rows[i] * stride + j] = __hip_atomic_load(
          srcs[i] + j, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  }
}

struct DpoGroup {
  int n;
  DpoCtx** cs;
  double** Xs;          // host arrays of fixed device pointers
  const double** nbrs;
  double** d_srcs;      // device array: each ctx's eval3
  int* d_rows;          // device array: output row per agent
  unsigned int** d_fin;   // device array: each ctx's F_EVAL_IN fence
  unsigned int** d_fout;  // device array: each ctx's F_EVAL_OUT fence
};

void dpo_group_destroy(void* g) {
  DpoGroup* gr = (DpoGroup*)g;
  if (!gr) return;
  dpo_ignore(hipFree(gr->d_srcs));
  dpo_ignore(hipFree(gr->d_rows));
  dpo_ignore(hipFree(gr->d_fin));
  dpo_ignore(hipFree(gr->d_fout));
  free(gr->cs); free(gr->Xs); free(gr->nbrs);
  delete gr;
}

void* dpo_group_create(void** handles, int n, double** Xs,
                       const double** nbrs, const int* rows) {
  if (n <= 0 || n > 256) return nullptr;
  DpoGroup* gr = new DpoGroup();
  gr->n = n;
  gr->cs = (DpoCtx**)malloc(n * sizeof(void*));
  gr->Xs = (double**)malloc(n * sizeof(void*));
  gr->nbrs = (const double**)malloc(n * sizeof(void*));
  double* srcs_h[256];
  unsigned int* fin_h[256];
  unsigned int* fout_h[256];
  for (int i = 0; i < n; ++i) {
    gr->cs[i] = (DpoCtx*)handles[i];
    gr->Xs[i] = Xs[i];
    gr->nbrs[i] = nbrs[i];
    srcs_h[i] = gr->cs[i]->eval3;
    fin_h[i] = gr->cs[i]->fences + F_EVAL_IN;
    fout_h[i] = gr->cs[i]->fences + F_EVAL_OUT;
  }
  DPO_CHECK(hipMalloc(&gr->d_srcs, n * sizeof(double*)));
  DPO_CHECK(hipMemcpy(gr->d_srcs, srcs_h, n * sizeof(double*),
                      hipMemcpyHostToDevice));
  DPO_CHECK(hipMalloc(&gr->d_rows, n * sizeof(int)));
  DPO_CHECK(hipMemcpy(gr->d_rows, rows, n * sizeof(int),
                      hipMemcpyHostToDevice));
  DPO_CHECK(hipMalloc(&gr->d_fin, n * sizeof(void*)));
  DPO_CHECK(hipMemcpy(gr->d_fin, fin_h, n * sizeof(void*),
                      hipMemcpyHostToDevice));
  DPO_CHECK(hipMalloc(&gr->d_fout, n * sizeof(void*)));
  DPO_CHECK(hipMemcpy(gr->d_fout, fout_h, n * sizeof(void*),
                      hipMemcpyHostToDevice));
  return gr;
}

// Fan-out solves for the active agents (indices into the group), on
// each agent's private exec stream, fenced against join_stream.
void dpo_group_solve_start(void* g, const int* ids, int k, double tol,
                           double Delta0, double accept_rho,
                           void* join_stream) {
  DpoGroup* gr = (DpoGroup*)g;
  for (int i = 0; i < k; ++i) {
    int a = ids[i];
    dpo_round_solve_async(gr->cs[a], gr->Xs[a], gr->nbrs[a], tol, Delta0,
                          accept_rho, join_stream);
  }
}

void dpo_group_solve_finish(void* g, const int* ids, int k,
                            int max_shrink, void* join_stream) {
  DpoGroup* gr = (DpoGroup*)g;
  for (int i = 0; i < k; ++i)
    dpo_round_solve_finish(gr->cs[ids[i]], max_shrink, nullptr,
                           join_stream);
}

// Fan-out evaluation of EVERY agent in the group into its eval3
// scratch (per-agent exec streams, fenced), then gather all rows into
// out_dev[rows[i]*row_stride .. +3] with one kernel on join_stream.
void dpo_group_eval(void* g, double* out_dev, long row_stride,
                    void* join_stream) {
  DpoGroup* gr = (DpoGroup*)g;
  hipStream_t js = (hipStream_t)join_stream;
  // one kernel signals every agent's IN fence (carries the boundary
  // scatter ordering); the per-agent eval graphs are then launched
  // bare on their exec streams — no event ceremony — and one gather
  // kernel on the join stream waits on & consumes every OUT fence.
  hipLaunchKernelGGL(k_fence_signal_many, dim3(1), dim3(256), 0, js,
                     (unsigned int* const*)gr->d_fin, gr->n);
  for (int i = 0; i < gr->n; ++i)
    round_eval_impl(gr->cs[i], gr->Xs[i], gr->nbrs[i], gr->cs[i]->eval3,
                    gr->cs[i]->exec_stream, /*wait_out=*/false);
  hipLaunchKernelGGL(k_gather3_fenced, dim3(1), dim3(256), 0, js,
                     out_dev, row_stride, gr->d_srcs, gr->d_rows,
                     (unsigned int* const*)gr->d_fout, gr->n);
}

}  // extern "C"
