// arrow_spmm.hip — MI355X (gfx950, CDNA4) kernels for the arrow-SpMM hot path.
//
// Built from scratch for CDNA4; replaces the reference's cupy/cuSPARSE CSRMM
// calls (spcl/arrow-matrix arrow_slim_mpi.py:158-244) and host permutation
// gathers (arrow_dec_mpi.py:421-437,526-544). See include/arrow_spmm.h for
// the ABI contract.
//
// Kernel design (CSR x tall-skinny dense, fp32, k = 1..128 typical):
//   * The path is HBM-bandwidth-bound (AI ~= 2.3 flop/byte), not a dense
//     contraction -> no MFMA. The levers are coalescing and load balance.
//   * k is mapped across lanes of a 64-wide wavefront in GROUP-lane
//     "row groups", each lane loading VEC consecutive floats of an X row
//     (float4 when k % 4 == 0) -> one fully-coalesced 64..512 B read per
//     touched X row, and coalesced C writes.
//   * Load balance on power-law rows (the first block-row A_0i holds the
//     hub vertices): rows are pre-split at upload time into work items of
//     at most SEG_NNZ nonzeros; split rows accumulate into C with
//     global fp32 atomics, whole rows write directly. Every row gets an
//     item, so beta=0 also zeroes empty rows.
//   * A's (col, val) stream is read once per item by all lanes of the
//     group (same-address broadcast within the wave's transaction).
//   * Work items are consumed either by a grid-stride loop (small
//     structures) or by the per-XCD queue scheduler (GROUP >= 4 and
//     >= 32M nnz by default): 8 contiguous nnz-balanced item segments
//     drained via per-XCD atomic chunk counters so each XCD walks one
//     tight X window in order (measured +54 % at k=128). Grabs are per
//     WORKGROUP at GROUP >= 8 (spmm_kernel_q) and per WAVE at GROUP < 8
//     (spmm_kernel_qw — no __syncthreads rendezvous; k=16 measured
//     +15 % over grid-stride, profiles/r02_ab2_sweep.log).

#include <hip/hip_runtime.h>
#include <algorithm>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <string>
#include <unordered_map>
#include <vector>

#include "../../include/arrow_spmm.h"

#define ARROW_ABI_VERSION 1001  // round 2: set_qblocks, create_opts flags bit 2, wave-grab scheduler

namespace {

thread_local std::string g_last_error;

void set_error(const std::string &msg) { g_last_error = msg; }

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      set_error(std::string(#expr) + ": " + hipGetErrorString(_e));            \
      return -1;                                                               \
    }                                                                          \
  } while (0)

constexpr int SEG_NNZ = 2048;     // max nonzeros per work item
constexpr int BLOCK_THREADS = 256;

struct CsrBlock {
  int64_t rows = 0, cols = 0, nnz = 0;
  // (col, val) packed 8-byte pairs (device, nnz). col < 0 encodes an index
  // into the SECOND feature matrix of arrow_spmm_dual: col = -(idx+1)
  // (the fused diagonal+first-block-column layout).
  int2 *pairs = nullptr;
  // work items (device): row (bit31 = atomic), [begin, end) into pairs
  int32_t *item_row = nullptr;
  int32_t *item_begin = nullptr;
  int32_t *item_end = nullptr;
  int64_t n_items = 0;
  int32_t *split_rows = nullptr;  // device: rows needing pre-zero at beta=0
  int64_t n_split_rows = 0;
  // XCD-contiguous item remap for this structure (uniform banded rows only:
  // on hub-heavy structures it serialises the heavy head onto one XCD)
  int xcd_remap = 0;
  // per-XCD queue scheduler: -1 follow ARROW_QUEUE env, 0 off, 1 on
  int queue_mode = -1;
  int64_t *qseg = nullptr;  // device, 9 nnz-balanced item-segment bounds
  int32_t *qctr = nullptr;  // device, 8 chunk counters padded 32 ints apart
  int q_blocks = 0;         // per-structure grid override (0 = ARROW_Q_BLOCKS)
};

int env_queue_default() {
  // -1 unset (per-launch policy: queue when GROUP >= 8 — measured
  // +54 % at k=128/cfg4, +18 % at k=64, +2 % at k=32 same-box, but
  // -19 % at k=16 where the per-chunk barrier dominates 4-lane groups;
  // profiles/r01_queue_chunk_sweep.txt, r01_ksweep_queue_20M.txt);
  // ARROW_QUEUE=0 forces grid-stride, ARROW_QUEUE=1 forces queues.
  static const int v = [] {
    const char *e = getenv("ARROW_QUEUE");
    if (!e || !e[0]) return -1;
    return (e[0] == '0') ? 0 : 1;
  }();
  return v;
}

std::unordered_map<int64_t, CsrBlock> g_blocks;
int64_t g_next_handle = 1;

// ---------------------------------------------------------------------------
// SpMM kernel.  GROUP lanes x VEC floats cover min(k, GROUP*VEC) columns;
// wider k is handled by a column-offset loop over launches (col_off).
// The per-item body is shared by the two schedulers below (grid-stride and
// per-XCD queue) — spmm_process_item computes one work item end to end.
// ---------------------------------------------------------------------------

template <int VEC, int GROUP, int BETA, bool GUARD>
__device__ __forceinline__ void spmm_process_item(
    const int2 *__restrict__ pairs, int32_t row_raw, int32_t b, int32_t e,
    const float *__restrict__ X0, const float *__restrict__ X1,
    float *__restrict__ C, int64_t k, int64_t col0, bool active,
    int lane_in_group, int nt_mode) {
  {
    const int32_t row = row_raw & 0x7fffffff;
    const bool is_split = row_raw < 0;

    float acc[VEC];
#pragma unroll
    for (int j = 0; j < VEC; ++j) acc[j] = 0.0f;

    // Stage GROUP (col,val) pairs cooperatively (ONE coalesced 8-B/lane
    // load per GROUP nonzeros), then broadcast them across the group with
    // wave shuffles — the A stream costs 1/GROUP memory instructions per
    // nonzero instead of 2.
    auto xaddr = [&](int2 mine, int u) {
      const int32_t c = __shfl(mine.x, u, GROUP);
      return (c < 0 ? X1 + (int64_t)(-c - 1) * k : X0 + (int64_t)c * k) + col0;
    };
    auto val = [&](int2 mine, int u) {
      return __int_as_float(__shfl(mine.y, u, GROUP));
    };
    auto consume = [&](int2 mine, int u) {
      const float *xr = xaddr(mine, u);
      const float v = val(mine, u);
      if (active) {
        if constexpr (VEC == 4) {
          const float4 xv = *reinterpret_cast<const float4 *>(xr);
          acc[0] = fmaf(v, xv.x, acc[0]);
          acc[1] = fmaf(v, xv.y, acc[1]);
          acc[2] = fmaf(v, xv.z, acc[2]);
          acc[3] = fmaf(v, xv.w, acc[3]);
        } else if constexpr (VEC == 2) {
          const float2 xv = *reinterpret_cast<const float2 *>(xr);
          acc[0] = fmaf(v, xv.x, acc[0]);
          acc[1] = fmaf(v, xv.y, acc[1]);
        } else {
          acc[0] = fmaf(v, xr[0], acc[0]);
        }
      }
    };
    // 4 entries at a time with the four X-row loads issued back to back
    // BEFORE any use: the rolled one-at-a-time loop stalls on each load's
    // s_waitcnt before the next can issue (in-order issue), which makes
    // short power-law rows latency-bound.
    auto consume4 = [&](int2 mine, int u) {
      if constexpr (VEC == 4) {
        const float *p0 = xaddr(mine, u + 0);
        const float *p1 = xaddr(mine, u + 1);
        const float *p2 = xaddr(mine, u + 2);
        const float *p3 = xaddr(mine, u + 3);
        const float v0 = val(mine, u + 0), v1 = val(mine, u + 1);
        const float v2 = val(mine, u + 2), v3 = val(mine, u + 3);
        if (active) {
          const float4 a0 = *reinterpret_cast<const float4 *>(p0);
          const float4 a1 = *reinterpret_cast<const float4 *>(p1);
          const float4 a2 = *reinterpret_cast<const float4 *>(p2);
          const float4 a3 = *reinterpret_cast<const float4 *>(p3);
          acc[0] = fmaf(v0, a0.x, acc[0]); acc[1] = fmaf(v0, a0.y, acc[1]);
          acc[2] = fmaf(v0, a0.z, acc[2]); acc[3] = fmaf(v0, a0.w, acc[3]);
          acc[0] = fmaf(v1, a1.x, acc[0]); acc[1] = fmaf(v1, a1.y, acc[1]);
          acc[2] = fmaf(v1, a1.z, acc[2]); acc[3] = fmaf(v1, a1.w, acc[3]);
          acc[0] = fmaf(v2, a2.x, acc[0]); acc[1] = fmaf(v2, a2.y, acc[1]);
          acc[2] = fmaf(v2, a2.z, acc[2]); acc[3] = fmaf(v2, a2.w, acc[3]);
          acc[0] = fmaf(v3, a3.x, acc[0]); acc[1] = fmaf(v3, a3.y, acc[1]);
          acc[2] = fmaf(v3, a3.z, acc[2]); acc[3] = fmaf(v3, a3.w, acc[3]);
        }
      } else {
        consume(mine, u); consume(mine, u + 1);
        consume(mine, u + 2); consume(mine, u + 3);
      }
    };
    // 8 loads in flight: a deg-8 power-law row issues its entire X-row
    // load set before any use (the 4-deep batch still serialises pairs of
    // batches on in-order issue)
    auto consume8 = [&](int2 mine, int u) {
      if constexpr (VEC == 4 && GROUP >= 8) {
        const float *p[8];
        float v[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          p[j] = xaddr(mine, u + j);
          v[j] = val(mine, u + j);
        }
        if (active) {
          float4 a[8];
#pragma unroll
          for (int j = 0; j < 8; ++j)
            a[j] = *reinterpret_cast<const float4 *>(p[j]);
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            acc[0] = fmaf(v[j], a[j].x, acc[0]);
            acc[1] = fmaf(v[j], a[j].y, acc[1]);
            acc[2] = fmaf(v[j], a[j].z, acc[2]);
            acc[3] = fmaf(v[j], a[j].w, acc[3]);
          }
        }
      } else {
        consume4(mine, u);
        consume4(mine, u + 4);
      }
    };
    // nontemporal A-stream load (each pair is read once by one wave): the
    // builtin wants a native vector type, HIP's int2 is a class
    typedef int nt_int2 __attribute__((ext_vector_type(2)));
    auto load_pair = [&](int32_t idx) -> int2 {
      if (nt_mode) {
        const nt_int2 raw = __builtin_nontemporal_load(
            reinterpret_cast<const nt_int2 *>(pairs + idx));
        return int2{raw.x, raw.y};
      }
      return pairs[idx];
    };
    int32_t base = b;
    for (; base + GROUP <= e; base += GROUP) {  // full chunks
      const int2 mine = load_pair(base + lane_in_group);
      if constexpr (GROUP >= 8) {
#pragma unroll
        for (int u = 0; u < GROUP; u += 8) consume8(mine, u);
      } else if constexpr (GROUP >= 4) {
#pragma unroll
        for (int u = 0; u < GROUP; u += 4) consume4(mine, u);
      } else {
#pragma unroll
        for (int u = 0; u < GROUP; ++u) consume(mine, u);
      }
    }
    if (base < e) {  // remainder (< GROUP entries)
      const int32_t t = base + lane_in_group;
      const int2 mine = (t < e) ? pairs[t] : int2{0, 0};
      const int cnt = e - base;
      int u = 0;
      if constexpr (GROUP >= 8) {
        for (; u + 8 <= cnt; u += 8) consume8(mine, u);
      }
      for (; u + 4 <= cnt; u += 4) consume4(mine, u);
      for (; u < cnt; ++u) consume(mine, u);
    }

    if (active) {
      float *cr = C + (int64_t)row * k + col0;
      if (is_split) {
#pragma unroll
        for (int j = 0; j < VEC; ++j) atomicAdd(cr + j, acc[j]);
      } else if constexpr (BETA == 1) {
        if constexpr (VEC == 4) {
          float4 old = *reinterpret_cast<float4 *>(cr);
          old.x += acc[0]; old.y += acc[1]; old.z += acc[2]; old.w += acc[3];
          *reinterpret_cast<float4 *>(cr) = old;
        } else {
#pragma unroll
          for (int j = 0; j < VEC; ++j) cr[j] += acc[j];
        }
      } else {
        if constexpr (VEC == 4) {
          if (nt_mode) {
            typedef float nt_float4 __attribute__((ext_vector_type(4)));
            const nt_float4 outv = {acc[0], acc[1], acc[2], acc[3]};
            __builtin_nontemporal_store(outv,
                                        reinterpret_cast<nt_float4 *>(cr));
          } else {
            *reinterpret_cast<float4 *>(cr) = float4{acc[0], acc[1], acc[2],
                                                     acc[3]};
          }
        } else {
#pragma unroll
          for (int j = 0; j < VEC; ++j) cr[j] = acc[j];
        }
      }
    }
  }
}

template <int VEC, int GROUP, int BETA, bool GUARD>
__global__ __launch_bounds__(BLOCK_THREADS) void spmm_kernel(
    const int2 *__restrict__ pairs,
    const int32_t *__restrict__ item_row, const int32_t *__restrict__ item_begin,
    const int32_t *__restrict__ item_end, int64_t n_items,
    const float *__restrict__ X0, const float *__restrict__ X1,
    float *__restrict__ C, int64_t k, int64_t col_off, int xcd_remap,
    int nt_mode) {
  constexpr int GROUPS_PER_BLOCK = BLOCK_THREADS / GROUP;
  const int lane_in_group = threadIdx.x % GROUP;
  const int group_in_block = threadIdx.x / GROUP;
  const int64_t col0 = col_off + (int64_t)lane_in_group * VEC;
  const bool active = !GUARD || (col0 < k);

  // XCD-aware workgroup remap (performance only): the dispatcher places
  // block b on XCD b%8, so remap block ids to give each XCD a CONTIGUOUS
  // range of work items — consecutive rows of a banded block then share the
  // XCD's private L2 window instead of interleaving across all 8 L2s.
  // Bijective form (cdna_hip_programming.md §XCD swizzle).
  int wg = blockIdx.x;
  if (xcd_remap) {
    const int nwg = gridDim.x;
    const int q = nwg / 8, rm = nwg % 8;
    const int xcd = blockIdx.x % 8, pos = blockIdx.x / 8;
    wg = (xcd < rm ? xcd * (q + 1) : rm * (q + 1) + (xcd - rm) * q) + pos;
  }

  int64_t item = (int64_t)wg * GROUPS_PER_BLOCK + group_in_block;
  const int64_t stride = (int64_t)gridDim.x * GROUPS_PER_BLOCK;

  // software prefetch of the NEXT item's metadata: without it every ~8-nnz
  // item pays a 3-deep dependent load chain (meta -> pairs -> X row) that
  // dominates short power-law rows
  int32_t next_row = 0, next_b = 0, next_e = 0;
  if (item < n_items) {
    next_row = __builtin_nontemporal_load(&item_row[item]);
    next_b = __builtin_nontemporal_load(&item_begin[item]);
    next_e = __builtin_nontemporal_load(&item_end[item]);
  }

  for (; item < n_items; item += stride) {
    const int32_t row_raw = next_row;
    const int32_t b = next_b;
    const int32_t e = next_e;
    const int64_t nxt = item + stride;
    if (nxt < n_items) {
      next_row = __builtin_nontemporal_load(&item_row[nxt]);
      next_b = __builtin_nontemporal_load(&item_begin[nxt]);
      next_e = __builtin_nontemporal_load(&item_end[nxt]);
    }
    spmm_process_item<VEC, GROUP, BETA, GUARD>(
        pairs, row_raw, b, e, X0, X1, C, k, col0, active, lane_in_group,
        nt_mode);
  }
}

// Per-XCD queue scheduler. The static grid-stride schedule above interleaves
// consecutive work items across all 8 XCDs and lets workgroups DRIFT apart
// over millions of items, so the ~10 consumers of each 512-B X row (rows
// within ±band) hit different L2s at different times — measured 3.2x X
// re-fetch, 16 % L2 hit (profiles/r01_pmc_sq_tcc_20M.txt). Here items are
// split into 8 CONTIGUOUS nnz-balanced row segments; each workgroup drains
// the segment of the XCD it actually runs on (HW_REG_XCC_ID, speed-only)
// via an atomic chunk counter, so every XCD walks one tight row window in
// order: consumers of an X row share one private L2, and the in-flight
// window stays a few MB. Workgroups that exhaust their queue steal from the
// next (placement-independent correctness; the register read is only a
// locality hint — cdna_hip_programming.md §Guideline 16).
__device__ __forceinline__ unsigned arrow_xcc_id() {
  // s_getreg_b32 HW_REG_XCC_ID: reg 20, offset 0, size 4 (gfx950)
  return __builtin_amdgcn_s_getreg((3 << 11) | (0 << 6) | 20);
}

template <int VEC, int GROUP, int BETA, bool GUARD>
__global__ __launch_bounds__(BLOCK_THREADS) void spmm_kernel_q(
    const int2 *__restrict__ pairs,
    const int32_t *__restrict__ item_row, const int32_t *__restrict__ item_begin,
    const int32_t *__restrict__ item_end,
    const int64_t *__restrict__ qseg,  // 9 segment bounds (items)
    int32_t *__restrict__ qctr,       // 8 counters, padded 32 ints apart
    int chunk_items,
    const float *__restrict__ X0, const float *__restrict__ X1,
    float *__restrict__ C, int64_t k, int64_t col_off, int nt_mode) {
  constexpr int GROUPS_PER_BLOCK = BLOCK_THREADS / GROUP;
  const int lane_in_group = threadIdx.x % GROUP;
  const int group_in_block = threadIdx.x / GROUP;
  const int64_t col0 = col_off + (int64_t)lane_in_group * VEC;
  const bool active = !GUARD || (col0 < k);
  __shared__ int64_t s_base;

  const unsigned q0 = arrow_xcc_id() & 7;
  for (unsigned qi = 0; qi < 8; ++qi) {
    const unsigned q = (q0 + qi) & 7;
    const int64_t lo = qseg[q], hi = qseg[q + 1];
    if (lo >= hi) continue;
    for (;;) {
      __syncthreads();
      if (threadIdx.x == 0) {
        s_base = lo + (int64_t)atomicAdd(&qctr[q * 32], chunk_items);
      }
      __syncthreads();
      const int64_t base = s_base;
      if (base >= hi) break;
      const int64_t end = base + chunk_items < hi ? base + chunk_items : hi;
      int64_t item = base + group_in_block;
      int32_t next_row = 0, next_b = 0, next_e = 0;
      if (item < end) {
        next_row = __builtin_nontemporal_load(&item_row[item]);
        next_b = __builtin_nontemporal_load(&item_begin[item]);
        next_e = __builtin_nontemporal_load(&item_end[item]);
      }
      for (; item < end; item += GROUPS_PER_BLOCK) {
        const int32_t row_raw = next_row;
        const int32_t b = next_b;
        const int32_t e = next_e;
        const int64_t nxt = item + GROUPS_PER_BLOCK;
        if (nxt < end) {
          next_row = __builtin_nontemporal_load(&item_row[nxt]);
          next_b = __builtin_nontemporal_load(&item_begin[nxt]);
          next_e = __builtin_nontemporal_load(&item_end[nxt]);
        }
        spmm_process_item<VEC, GROUP, BETA, GUARD>(
            pairs, row_raw, b, e, X0, X1, C, k, col0, active, lane_in_group,
            nt_mode);
      }
    }
  }
}

// Per-WAVE queue grab variant of spmm_kernel_q: each 64-lane wave pulls its
// own chunk from the XCD's counter (lane-0 atomicAdd broadcast by wave
// shuffle) — no __syncthreads per grab. Removes the whole-block rendezvous
// that made queues lose at small GROUP (k=16: 16 groups per wave stall on
// the slowest group of all 4 waves), at the cost of 4x the atomic rate
// (fine: per-XCD counters, dequeue ~0.25-1.1 us uncontended and these are
// thousands of grabs per segment, MI355X_MICROARCH.md §dequeue).
template <int VEC, int GROUP, int BETA, bool GUARD>
__global__ __launch_bounds__(BLOCK_THREADS) void spmm_kernel_qw(
    const int2 *__restrict__ pairs,
    const int32_t *__restrict__ item_row, const int32_t *__restrict__ item_begin,
    const int32_t *__restrict__ item_end,
    const int64_t *__restrict__ qseg,  // 9 segment bounds (items)
    int32_t *__restrict__ qctr,       // 8 counters, padded 32 ints apart
    int chunk_items,                  // per-WAVE grab size (items)
    const float *__restrict__ X0, const float *__restrict__ X1,
    float *__restrict__ C, int64_t k, int64_t col_off, int nt_mode) {
  constexpr int GROUPS_PER_WAVE = 64 / GROUP;
  const int lane = threadIdx.x & 63;
  const int lane_in_group = threadIdx.x % GROUP;
  const int group_in_wave = lane / GROUP;
  const int64_t col0 = col_off + (int64_t)lane_in_group * VEC;
  const bool active = !GUARD || (col0 < k);

  const unsigned q0 = arrow_xcc_id() & 7;
  for (unsigned qi = 0; qi < 8; ++qi) {
    const unsigned q = (q0 + qi) & 7;
    const int64_t lo = qseg[q], hi = qseg[q + 1];
    if (lo >= hi) continue;
    for (;;) {
      int grab = 0;
      if (lane == 0) grab = atomicAdd(&qctr[q * 32], chunk_items);
      grab = __shfl(grab, 0, 64);
      const int64_t base = lo + grab;
      if (base >= hi) break;
      const int64_t end = base + chunk_items < hi ? base + chunk_items : hi;
      int64_t item = base + group_in_wave;
      int32_t next_row = 0, next_b = 0, next_e = 0;
      if (item < end) {
        next_row = __builtin_nontemporal_load(&item_row[item]);
        next_b = __builtin_nontemporal_load(&item_begin[item]);
        next_e = __builtin_nontemporal_load(&item_end[item]);
      }
      for (; item < end; item += GROUPS_PER_WAVE) {
        const int32_t row_raw = next_row;
        const int32_t b = next_b;
        const int32_t e = next_e;
        const int64_t nxt = item + GROUPS_PER_WAVE;
        if (nxt < end) {
          next_row = __builtin_nontemporal_load(&item_row[nxt]);
          next_b = __builtin_nontemporal_load(&item_begin[nxt]);
          next_e = __builtin_nontemporal_load(&item_end[nxt]);
        }
        spmm_process_item<VEC, GROUP, BETA, GUARD>(
            pairs, row_raw, b, e, X0, X1, C, k, col0, active, lane_in_group,
            nt_mode);
      }
    }
  }
}

__global__ void zero_rows_kernel(float *__restrict__ C,
                                 const int32_t *__restrict__ rows,
                                 int64_t n_rows, int64_t k) {
  const int64_t total = n_rows * k;
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (int64_t)gridDim.x * blockDim.x) {
    const int64_t i = t / k, j = t % k;
    C[(int64_t)rows[i] * k + j] = 0.0f;
  }
}

// ---------------------------------------------------------------------------
// Row gather / scatter / scatter-add (permutation routing on-device).
// ---------------------------------------------------------------------------

enum class RouteOp { Gather, Scatter, ScatterAdd };

template <typename VT, RouteOp OP>
__global__ void route_rows_kernel(float *__restrict__ dst,
                                  const float *__restrict__ src,
                                  const int64_t *__restrict__ idx, int64_t n,
                                  int64_t k_vec) {
  // k_vec = k / (elements per VT); row r maps dst<->src[idx[r]]
  const int64_t total = n * k_vec;
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (int64_t)gridDim.x * blockDim.x) {
    const int64_t i = t / k_vec, j = t % k_vec;
    if constexpr (OP == RouteOp::Gather) {
      reinterpret_cast<VT *>(dst)[i * k_vec + j] =
          reinterpret_cast<const VT *>(src)[idx[i] * k_vec + j];
    } else if constexpr (OP == RouteOp::Scatter) {
      reinterpret_cast<VT *>(dst)[idx[i] * k_vec + j] =
          reinterpret_cast<const VT *>(src)[i * k_vec + j];
    } else {
      const VT v = reinterpret_cast<const VT *>(src)[i * k_vec + j];
      VT *d = reinterpret_cast<VT *>(dst) + idx[i] * k_vec + j;
      if constexpr (sizeof(VT) == 4) {
        *d += v;
      } else {
        const float4 a = *reinterpret_cast<const float4 *>(&v);
        float4 b = *reinterpret_cast<float4 *>(d);
        b.x += a.x; b.y += a.y; b.z += a.z; b.w += a.w;
        *reinterpret_cast<float4 *>(d) = b;
      }
    }
  }
}

// dst[dst_idx[i], :] (+)= src[src_idx[i], :] — fused gather+scatter for the
// rank-local share of the permutation exchange (one read + one write per
// element instead of a gather pass plus a scatter pass).
template <typename VT, bool ADD>
__global__ void permute_rows_kernel(float *__restrict__ dst,
                                    const float *__restrict__ src,
                                    const int64_t *__restrict__ dst_idx,
                                    const int64_t *__restrict__ src_idx,
                                    int64_t n, int64_t k_vec) {
  const int64_t total = n * k_vec;
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (int64_t)gridDim.x * blockDim.x) {
    const int64_t i = t / k_vec, j = t % k_vec;
    const VT v = reinterpret_cast<const VT *>(src)[src_idx[i] * k_vec + j];
    VT *d = reinterpret_cast<VT *>(dst) + dst_idx[i] * k_vec + j;
    if constexpr (!ADD) {
      *d = v;
    } else if constexpr (sizeof(VT) == 4) {
      *d += v;
    } else {
      const float4 a = *reinterpret_cast<const float4 *>(&v);
      float4 b = *reinterpret_cast<float4 *>(d);
      b.x += a.x; b.y += a.y; b.z += a.z; b.w += a.w;
      *reinterpret_cast<float4 *>(d) = b;
    }
  }
}

int launch_permute(int add, float *dst, const float *src, const int64_t *dst_idx,
                   const int64_t *src_idx, int64_t n, int64_t k,
                   hipStream_t stream) {
  if (n == 0) return 0;
  const bool vec4 = (k % 4 == 0);
  const int64_t k_vec = vec4 ? k / 4 : k;
  const int threads = 256;
  int blocks = (int)std::min<int64_t>((n * k_vec + threads - 1) / threads, 16384);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(threads), 0, stream, dst, src,
                       dst_idx, src_idx, n, k_vec);
  };
  if (vec4) {
    if (add) launch(permute_rows_kernel<float4, true>);
    else     launch(permute_rows_kernel<float4, false>);
  } else {
    if (add) launch(permute_rows_kernel<float, true>);
    else     launch(permute_rows_kernel<float, false>);
  }
  HIP_CHECK(hipGetLastError());
  return 0;
}

int launch_route(RouteOp op, float *dst, const float *src, const int64_t *idx,
                 int64_t n, int64_t k, hipStream_t stream) {
  if (n == 0) return 0;
  const bool vec4 = (k % 4 == 0);
  const int64_t k_vec = vec4 ? k / 4 : k;
  const int64_t total = n * k_vec;
  const int threads = 256;
  int blocks = (int)std::min<int64_t>((total + threads - 1) / threads, 16384);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(threads), 0, stream, dst, src,
                       idx, n, k_vec);
  };
  if (vec4) {
    switch (op) {
      case RouteOp::Gather: launch(route_rows_kernel<float4, RouteOp::Gather>); break;
      case RouteOp::Scatter: launch(route_rows_kernel<float4, RouteOp::Scatter>); break;
      case RouteOp::ScatterAdd: launch(route_rows_kernel<float4, RouteOp::ScatterAdd>); break;
    }
  } else {
    switch (op) {
      case RouteOp::Gather: launch(route_rows_kernel<float, RouteOp::Gather>); break;
      case RouteOp::Scatter: launch(route_rows_kernel<float, RouteOp::Scatter>); break;
      case RouteOp::ScatterAdd: launch(route_rows_kernel<float, RouteOp::ScatterAdd>); break;
    }
  }
  HIP_CHECK(hipGetLastError());
  return 0;
}

// ---------------------------------------------------------------------------
// SpMM dispatch over (VEC, GROUP, BETA, GUARD)
// ---------------------------------------------------------------------------

struct LaunchCfg {
  int vec;
  int group;
  int64_t col_span;  // columns covered per launch = group * vec
};

LaunchCfg pick_cfg(int64_t k) {
  static const int g64 =
      [] { const char *e = getenv("ARROW_SPMM_G64");
           return (e && e[0] == '1') ? 1 : 0; }();
  if (g64 && k == 128) return {2, 64, 128};  // A/B: wave-wide groups, float2
  static const int k16g8 =
      [] { const char *e = getenv("ARROW_K16_G8");
           return (e && e[0] == '1') ? 1 : 0; }();
  if (k16g8 && k == 16) return {2, 8, 16};  // A/B: float2 x 8 lanes,
                                            // block-grab queue eligible
  int vec = (k % 4 == 0) ? 4 : (k % 2 == 0) ? 2 : 1;
  int64_t lanes_needed = (k + vec - 1) / vec;
  int group = 1;
  while (group < lanes_needed && group < 64) group <<= 1;
  return {vec, group, (int64_t)group * vec};
}

template <int VEC, int GROUP>
int launch_spmm_vg(const CsrBlock &blk, const float *X0, const float *X1,
                   float *C, int64_t k, int beta, hipStream_t stream) {
  constexpr int GROUPS_PER_BLOCK = BLOCK_THREADS / GROUP;
  // Non-temporal pairs load + C store: default ON (+1.4-1.7 % at cfg4
  // with the queue scheduler — the once-read A stream and once-written C
  // stop evicting the X window; profiles/r01_nt_chunk_k32_ab.txt,
  // r01_colsort_nt_ab.txt). ARROW_SPMM_NT=0 restores cached accesses.
  static const int nt_mode = [] {
    const char *e = getenv("ARROW_SPMM_NT");
    return (e && e[0] == '0') ? 0 : 1;
  }();
  const int qd = blk.queue_mode >= 0 ? blk.queue_mode : env_queue_default();
  // Per-wave queue grabs (spmm_kernel_qw): no per-chunk block barrier.
  // Default AUTO: wave variant at GROUP < 8 (k=16: measured +10% over the
  // fused grid-stride and +15% over the round-1 baseline,
  // profiles/r02_ab2_sweep.log k16_fused_q1_qw), block variant at
  // GROUP >= 8 (k=128: wave measured -17%, r02_ab1). ARROW_QWAVE=0/1
  // forces.
  static const int qwave_env = [] {
    const char *e = getenv("ARROW_QWAVE");
    if (!e || !e[0]) return -1;
    return (e[0] == '1') ? 1 : 0;
  }();
  const bool qwave = qwave_env >= 0 ? (qwave_env == 1) : (GROUP < 8);
  // policy default: queues need GROUP >= 4 AND enough work per structure —
  // below ~32M nnz (ARROW_Q_MIN_NNZ overrides) the per-grab overhead
  // outweighs the L2 window benefit
  static const int64_t q_min_nnz = [] {
    const char *e = getenv("ARROW_Q_MIN_NNZ");
    return e ? (int64_t)atoll(e) : (32LL << 20);
  }();
  const bool useq = blk.qseg &&
      (qd >= 0 ? qd : (GROUP >= 4 && blk.nnz >= q_min_nnz));
  // Queue mode: size the grid to residency (8 blocks/CU fit at this
  // occupancy — 4 waves/WG, 8 waves/SIMD), not to the item count; chunk =
  // 4 rounds per grab (measured best on the fused cfg4 structure: 1/2/4 →
  // 4306/4450/4552 GF/s, profiles/r02_ab2_sweep.log).
  static const int q_chunk_mult = [] {
    const char *e = getenv("ARROW_Q_CHUNK");
    return e ? std::max(1, atoi(e)) : 4;
  }();
  static const int q_blocks_env = [] {
    const char *e = getenv("ARROW_Q_BLOCKS");
    return e ? std::max(8, atoi(e)) : 2048;
  }();
  const int q_blocks = blk.q_blocks > 0 ? blk.q_blocks : q_blocks_env;
  const int chunk_items = GROUPS_PER_BLOCK * q_chunk_mult;
  // per-wave grabs pull GROUPS_PER_WAVE-item rounds. Default 2 rounds —
  // the wave variant regresses at larger grabs (k=16: 1811 GF/s at 2 vs
  // 1342 at 4, 786 at 8 — its locality window is per-wave, so bigger
  // chunks smear it). ARROW_QW_CHUNK overrides.
  static const int qw_chunk_mult = [] {
    const char *e = getenv("ARROW_QW_CHUNK");
    return e ? std::max(1, atoi(e)) : 2;
  }();
  const int chunk_items_w = (64 / GROUP) * qw_chunk_mult;
  int blocks = (int)std::min<int64_t>(
      (blk.n_items + GROUPS_PER_BLOCK - 1) / GROUPS_PER_BLOCK,
      useq ? q_blocks : 8192);
  if (blocks < 1) blocks = 1;
  const int64_t span = (int64_t)GROUP * VEC;
  for (int64_t col_off = 0; col_off < k; col_off += span) {
    const bool guard = (col_off + span > k);
    auto run = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(blocks), dim3(BLOCK_THREADS), 0, stream,
                         blk.pairs, blk.item_row, blk.item_begin,
                         blk.item_end, blk.n_items, X0, X1, C, k, col_off,
                         blk.xcd_remap, nt_mode);
    };
    auto runq = [&](auto kern, int chunk) {
      hipLaunchKernelGGL(kern, dim3(blocks), dim3(BLOCK_THREADS), 0, stream,
                         blk.pairs, blk.item_row, blk.item_begin,
                         blk.item_end, blk.qseg, blk.qctr, chunk,
                         X0, X1, C, k, col_off, nt_mode);
    };
    if (useq) {
      HIP_CHECK(hipMemsetAsync(blk.qctr, 0, 8 * 32 * sizeof(int32_t), stream));
      if (qwave) {
        if (beta == 0) {
          if (guard) runq(spmm_kernel_qw<VEC, GROUP, 0, true>, chunk_items_w);
          else       runq(spmm_kernel_qw<VEC, GROUP, 0, false>, chunk_items_w);
        } else {
          if (guard) runq(spmm_kernel_qw<VEC, GROUP, 1, true>, chunk_items_w);
          else       runq(spmm_kernel_qw<VEC, GROUP, 1, false>, chunk_items_w);
        }
      } else if (beta == 0) {
        if (guard) runq(spmm_kernel_q<VEC, GROUP, 0, true>, chunk_items);
        else       runq(spmm_kernel_q<VEC, GROUP, 0, false>, chunk_items);
      } else {
        if (guard) runq(spmm_kernel_q<VEC, GROUP, 1, true>, chunk_items);
        else       runq(spmm_kernel_q<VEC, GROUP, 1, false>, chunk_items);
      }
    } else if (beta == 0) {
      if (guard) run(spmm_kernel<VEC, GROUP, 0, true>);
      else       run(spmm_kernel<VEC, GROUP, 0, false>);
    } else {
      if (guard) run(spmm_kernel<VEC, GROUP, 1, true>);
      else       run(spmm_kernel<VEC, GROUP, 1, false>);
    }
    HIP_CHECK(hipGetLastError());
  }
  return 0;
}

template <int VEC>
int launch_spmm_v(const CsrBlock &blk, const float *X0, const float *X1,
                  float *C, int64_t k, int beta, int group, hipStream_t stream) {
  switch (group) {
    case 1:  return launch_spmm_vg<VEC, 1>(blk, X0, X1, C, k, beta, stream);
    case 2:  return launch_spmm_vg<VEC, 2>(blk, X0, X1, C, k, beta, stream);
    case 4:  return launch_spmm_vg<VEC, 4>(blk, X0, X1, C, k, beta, stream);
    case 8:  return launch_spmm_vg<VEC, 8>(blk, X0, X1, C, k, beta, stream);
    case 16: return launch_spmm_vg<VEC, 16>(blk, X0, X1, C, k, beta, stream);
    case 32: return launch_spmm_vg<VEC, 32>(blk, X0, X1, C, k, beta, stream);
    default: return launch_spmm_vg<VEC, 64>(blk, X0, X1, C, k, beta, stream);
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// C ABI
// ---------------------------------------------------------------------------

extern "C" {

int arrow_abi_version(void) { return ARROW_ABI_VERSION; }

const char *arrow_last_error(void) { return g_last_error.c_str(); }

int arrow_device_count(void) {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

int arrow_set_device(int device) {
  HIP_CHECK(hipSetDevice(device));
  return 0;
}

int arrow_synchronize(void) {
  HIP_CHECK(hipDeviceSynchronize());
  return 0;
}

static int64_t csr_create_impl(int64_t rows, int64_t cols, int64_t nnz,
                               const int64_t *indptr, const int32_t *indices,
                               const float *data, const int64_t *row_ids,
                               int flags = 0) {
  if (rows < 0 || cols < 0 || nnz < 0 || (rows > 0 && !indptr)) {
    set_error("arrow_csr_create: bad arguments");
    return -1;
  }
  if (nnz > INT32_MAX) {
    set_error("arrow_csr_create: nnz exceeds int32 (per-block limit)");
    return -1;
  }
  CsrBlock blk;
  blk.rows = rows;
  blk.cols = cols;
  blk.nnz = nnz;

  // Build work items on the host: every row gets at least one item; rows
  // with > SEG_NNZ nonzeros are split and marked atomic (bit 31).
  std::vector<int32_t> item_row, item_begin, item_end, split_rows;
  item_row.reserve(rows + nnz / SEG_NNZ + 1);
  for (int64_t r = 0; r < rows; ++r) {
    const int64_t b = indptr[r], e = indptr[r + 1];
    // row_ids (optional) maps structure-row -> output C row, enabling
    // reordered layouts (e.g. the hub-sorted X_0 structure)
    const int32_t out_row = (int32_t)(row_ids ? row_ids[r] : r);
    if (e - b <= SEG_NNZ) {
      item_row.push_back(out_row);
      item_begin.push_back((int32_t)b);
      item_end.push_back((int32_t)e);
    } else {
      split_rows.push_back(out_row);
      for (int64_t s = b; s < e; s += SEG_NNZ) {
        item_row.push_back(out_row | INT32_MIN);
        item_begin.push_back((int32_t)s);
        item_end.push_back((int32_t)std::min<int64_t>(s + SEG_NNZ, e));
      }
    }
  }
  blk.n_items = (int64_t)item_row.size();
  blk.n_split_rows = (int64_t)split_rows.size();

  // ARROW_CSR_COL_ITEMS: order work items by their first column instead of
  // by row. For hub-heavy structures (the first block-row: long CSR rows
  // split into column-contiguous items) this makes a queue segment a
  // COLUMN window of X, so the ~concurrent hub-row streams share the
  // XCD's L2. Output correctness is order-independent (each non-split row
  // appears once; split rows accumulate atomically either way).
  if (flags & 1) {
    const int64_t n_it = (int64_t)item_row.size();
    std::vector<int64_t> order(n_it);
    for (int64_t i = 0; i < n_it; ++i) order[i] = i;
    std::stable_sort(order.begin(), order.end(),
                     [&](int64_t a, int64_t b) {
                       const int32_t ca = item_begin[a] < item_end[a]
                                              ? indices[item_begin[a]]
                                              : INT32_MAX;
                       const int32_t cb = item_begin[b] < item_end[b]
                                              ? indices[item_begin[b]]
                                              : INT32_MAX;
                       return ca < cb;
                     });
    std::vector<int32_t> r2(n_it), b2(n_it), e2(n_it);
    for (int64_t i = 0; i < n_it; ++i) {
      r2[i] = item_row[order[i]];
      b2[i] = item_begin[order[i]];
      e2[i] = item_end[order[i]];
    }
    item_row.swap(r2);
    item_begin.swap(b2);
    item_end.swap(e2);
  }

  // nnz-balanced contiguous item segments for the per-XCD queue scheduler
  // (items are in row order — or column order with flags&1 — so a segment
  // is a contiguous window of X)
  int64_t qseg_h[9];
  {
    // weight = nnz + 1 per item: balances by data volume while keeping
    // empty/short rows (fixed per-item overhead) spread across queues
    const int64_t n_it = blk.n_items;
    const int64_t total_w = nnz + n_it;
    int64_t cum = 0, target_q = 1;
    qseg_h[0] = 0;
    for (int64_t i = 0; i < n_it && target_q < 8; ++i) {
      cum += (item_end[(size_t)i] - item_begin[(size_t)i]) + 1;
      while (target_q < 8 && cum * 8 >= total_w * target_q) {
        qseg_h[target_q++] = i + 1;
      }
    }
    while (target_q <= 8) qseg_h[target_q++] = n_it;
  }

  // flags&2: TWO-LEVEL hub ordering — keep the 8 nnz-balanced ROW segments
  // (so each XCD still owns a contiguous C range and load stays balanced),
  // then sort the items WITHIN each segment by first column: an XCD's
  // queue walk becomes a column-window sweep of X while its C writes stay
  // inside the segment's row range. Order-independent for correctness
  // (each non-split row appears once; split rows accumulate atomically).
  if (flags & 2) {
    for (int q = 0; q < 8; ++q) {
      const int64_t lo = qseg_h[q], hi = qseg_h[q + 1];
      if (hi - lo < 2) continue;
      std::vector<int64_t> order((size_t)(hi - lo));
      for (int64_t i = lo; i < hi; ++i) order[(size_t)(i - lo)] = i;
      std::stable_sort(order.begin(), order.end(),
                       [&](int64_t a, int64_t b) {
                         const int32_t ca = item_begin[(size_t)a] < item_end[(size_t)a]
                                                ? indices[item_begin[(size_t)a]]
                                                : INT32_MAX;
                         const int32_t cb = item_begin[(size_t)b] < item_end[(size_t)b]
                                                ? indices[item_begin[(size_t)b]]
                                                : INT32_MAX;
                         return ca < cb;
                       });
      std::vector<int32_t> r2((size_t)(hi - lo)), b2((size_t)(hi - lo)),
          e2((size_t)(hi - lo));
      for (int64_t i = 0; i < hi - lo; ++i) {
        r2[(size_t)i] = item_row[(size_t)order[(size_t)i]];
        b2[(size_t)i] = item_begin[(size_t)order[(size_t)i]];
        e2[(size_t)i] = item_end[(size_t)order[(size_t)i]];
      }
      std::copy(r2.begin(), r2.end(), item_row.begin() + (size_t)lo);
      std::copy(b2.begin(), b2.end(), item_begin.begin() + (size_t)lo);
      std::copy(e2.begin(), e2.end(), item_end.begin() + (size_t)lo);
    }
  }

  // pack (col, val) into 8-byte pairs for the staged kernel loads
  std::vector<int2> pairs((size_t)nnz);
  for (int64_t t = 0; t < nnz; ++t) {
    pairs[(size_t)t].x = indices[t];
    float v = data[t];
    pairs[(size_t)t].y = *reinterpret_cast<const int32_t *>(&v);
  }

  auto upload = [&](void **dst, const void *src, size_t bytes) -> int {
    if (bytes == 0) { *dst = nullptr; return 0; }
    HIP_CHECK(hipMalloc(dst, bytes));
    HIP_CHECK(hipMemcpy(*dst, src, bytes, hipMemcpyHostToDevice));
    return 0;
  };
  if (upload((void **)&blk.pairs, pairs.data(), pairs.size() * sizeof(int2)) ||
      upload((void **)&blk.item_row, item_row.data(), item_row.size() * 4) ||
      upload((void **)&blk.item_begin, item_begin.data(), item_begin.size() * 4) ||
      upload((void **)&blk.item_end, item_end.data(), item_end.size() * 4) ||
      upload((void **)&blk.split_rows, split_rows.data(), split_rows.size() * 4) ||
      upload((void **)&blk.qseg, qseg_h, sizeof(qseg_h))) {
    return -1;
  }
  HIP_CHECK(hipMalloc((void **)&blk.qctr, 8 * 32 * sizeof(int32_t)));
  const int64_t h = g_next_handle++;
  g_blocks.emplace(h, blk);
  return h;
}

int64_t arrow_csr_create(int64_t rows, int64_t cols, int64_t nnz,
                         const int64_t *indptr, const int32_t *indices,
                         const float *data) {
  return csr_create_impl(rows, cols, nnz, indptr, indices, data, nullptr);
}

int64_t arrow_csr_create_rows(int64_t rows, int64_t cols, int64_t nnz,
                              const int64_t *indptr, const int32_t *indices,
                              const float *data, const int64_t *row_ids) {
  return csr_create_impl(rows, cols, nnz, indptr, indices, data, row_ids);
}

int64_t arrow_csr_create_opts(int64_t rows, int64_t cols, int64_t nnz,
                              const int64_t *indptr, const int32_t *indices,
                              const float *data, const int64_t *row_ids,
                              int flags) {
  return csr_create_impl(rows, cols, nnz, indptr, indices, data, row_ids,
                         flags);
}

int arrow_csr_destroy(int64_t handle) {
  auto it = g_blocks.find(handle);
  if (it == g_blocks.end()) {
    set_error("arrow_csr_destroy: bad handle");
    return -1;
  }
  CsrBlock &b = it->second;
  for (void *p : {(void *)b.pairs, (void *)b.item_row,
                  (void *)b.item_begin, (void *)b.item_end,
                  (void *)b.split_rows, (void *)b.qseg, (void *)b.qctr}) {
    if (p) (void)hipFree(p);
  }
  g_blocks.erase(it);
  return 0;
}

int64_t arrow_csr_nnz(int64_t handle) {
  auto it = g_blocks.find(handle);
  if (it == g_blocks.end()) return -1;
  return it->second.nnz;
}

int arrow_csr_set_xcd_remap(int64_t handle, int enable) {
  auto it = g_blocks.find(handle);
  if (it == g_blocks.end()) {
    set_error("arrow_csr_set_xcd_remap: bad handle");
    return -1;
  }
  it->second.xcd_remap = enable ? 1 : 0;
  return 0;
}

int arrow_csr_set_queue(int64_t handle, int mode) {
  auto it = g_blocks.find(handle);
  if (it == g_blocks.end()) {
    set_error("arrow_csr_set_queue: bad handle");
    return -1;
  }
  it->second.queue_mode = mode < 0 ? -1 : (mode ? 1 : 0);
  return 0;
}

int arrow_csr_set_qblocks(int64_t handle, int blocks) {
  auto it = g_blocks.find(handle);
  if (it == g_blocks.end()) {
    set_error("arrow_csr_set_qblocks: bad handle");
    return -1;
  }
  it->second.q_blocks = blocks > 0 ? blocks : 0;
  return 0;
}

int arrow_spmm_dual(int64_t handle, const float *X0_dev, const float *X1_dev,
                    float *C_dev, int64_t k, int beta, void *stream_v) {
  auto it = g_blocks.find(handle);
  if (it == g_blocks.end()) {
    set_error("arrow_spmm: bad handle");
    return -1;
  }
  if (k <= 0 || !X0_dev || !X1_dev || !C_dev) {
    set_error("arrow_spmm: bad arguments");
    return -1;
  }
  const CsrBlock &blk = it->second;
  hipStream_t stream = (hipStream_t)stream_v;

  // beta=0: split rows are accumulated with atomics, so pre-zero them.
  if (beta == 0 && blk.n_split_rows > 0) {
    const int64_t total = blk.n_split_rows * k;
    int blocks = (int)std::min<int64_t>((total + 255) / 256, 4096);
    hipLaunchKernelGGL(zero_rows_kernel, dim3(blocks), dim3(256), 0, stream,
                       C_dev, blk.split_rows, blk.n_split_rows, k);
    HIP_CHECK(hipGetLastError());
  }

  const LaunchCfg cfg = pick_cfg(k);
  switch (cfg.vec) {
    case 4: return launch_spmm_v<4>(blk, X0_dev, X1_dev, C_dev, k, beta, cfg.group, stream);
    case 2: return launch_spmm_v<2>(blk, X0_dev, X1_dev, C_dev, k, beta, cfg.group, stream);
    default: return launch_spmm_v<1>(blk, X0_dev, X1_dev, C_dev, k, beta, cfg.group, stream);
  }
}

int arrow_spmm(int64_t handle, const float *X_dev, float *C_dev, int64_t k,
               int beta, void *stream_v) {
  return arrow_spmm_dual(handle, X_dev, X_dev, C_dev, k, beta, stream_v);
}

int arrow_permute_rows_f32(float *dst, const float *src, const int64_t *dst_idx,
                           const int64_t *src_idx, int64_t n, int64_t k,
                           void *stream) {
  return launch_permute(0, dst, src, dst_idx, src_idx, n, k, (hipStream_t)stream);
}

int arrow_permute_add_rows_f32(float *dst, const float *src,
                               const int64_t *dst_idx, const int64_t *src_idx,
                               int64_t n, int64_t k, void *stream) {
  return launch_permute(1, dst, src, dst_idx, src_idx, n, k, (hipStream_t)stream);
}

int arrow_gather_rows_f32(const float *src, float *dst, const int64_t *idx,
                          int64_t n, int64_t k, void *stream) {
  return launch_route(RouteOp::Gather, dst, src, idx, n, k, (hipStream_t)stream);
}

int arrow_scatter_rows_f32(float *dst, const float *src, const int64_t *idx,
                           int64_t n, int64_t k, void *stream) {
  return launch_route(RouteOp::Scatter, dst, src, idx, n, k, (hipStream_t)stream);
}

int arrow_scatter_add_rows_f32(float *dst, const float *src, const int64_t *idx,
                               int64_t n, int64_t k, void *stream) {
  return launch_route(RouteOp::ScatterAdd, dst, src, idx, n, k, (hipStream_t)stream);
}

}  // extern "C"
