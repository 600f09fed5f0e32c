// kolibrie_amd native CDNA4 kernels (gfx950 / MI355X).
//
// MI355X-native implementations of the engine's hot loops:
//   K1 scan_probe  — per-binding-row index probe: binary-search range over a
//                    sorted packed-key permutation, two-pass count+emit.
//                    (replaces reference engine.rs:1018-1251 scan loops)
//   K2 hash_join   — chained open-addressing hash join over k int32 key
//                    columns, build+count+emit.
//                    (replaces engine.rs:1367 hash_join_solution_sequences
//                     and shared/src/join_algorithm.rs:64-265)
//   K5 filter      — postfix-bytecode predicate over id columns + the f64
//                    value column (replaces types.rs:373 evaluate_with_ids).
//
// Design notes (per CDNA4 guide): wave64, 256-thread blocks, grid-stride
// loops capped so the 256-CU chip is saturated without oversubscription;
// all data int32/int64 columns in HBM — memory-latency-bound probes rely on
// high occupancy; atomics are agent-scope (device) by default.
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdint>
#include <atomic>
#include <chrono>
#include <cmath>
#include <string_view>
#include <cstring>
#include <thread>
#include <unordered_map>

namespace py = pybind11;
#include <vector>

#define HIP_OK(expr)                                                          \
  do {                                                                        \
    hipError_t _e = (expr);                                                   \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));      \
  } while (0)

namespace {

constexpr int kBlock = 256;

inline int grid_for(int64_t n) {
  int64_t blocks = (n + kBlock - 1) / kBlock;
  // 256 CUs x 8 blocks/CU: cap and grid-stride the rest (guide G11)
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return static_cast<int>(blocks);
}

__device__ __forceinline__ int64_t lower_bound_i64(const int64_t* __restrict__ a,
                                                   int64_t n, int64_t key) {
  int64_t lo = 0, hi = n;
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (a[mid] < key) lo = mid + 1; else hi = mid;
  }
  return lo;
}

__device__ __forceinline__ int64_t upper_bound_i64(const int64_t* __restrict__ a,
                                                   int64_t n, int64_t key) {
  int64_t lo = 0, hi = n;
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (a[mid] <= key) lo = mid + 1; else hi = mid;
  }
  return lo;
}

// ---------------------------------------------------------------- K1: probe
// exact mode: probe key = packed (a,b) per row; range mode: probe the full
// range of leading component a (keys built from the int32 value).
//
__global__ void probe_count_exact(const int64_t* __restrict__ key12, int64_t n,
                                  const int64_t* __restrict__ keys, int64_t m,
                                  int64_t* __restrict__ lo_out,
                                  int32_t* __restrict__ cnt_out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < m;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t k = keys[i];
    int64_t lo = lower_bound_i64(key12, n, k);
    int64_t hi = upper_bound_i64(key12, n, k);
    lo_out[i] = lo;
    cnt_out[i] = static_cast<int32_t>(hi - lo);
  }
}

__global__ void probe_count_range(const int64_t* __restrict__ key12, int64_t n,
                                  const int32_t* __restrict__ vals, int64_t m,
                                  int64_t* __restrict__ lo_out,
                                  int32_t* __restrict__ cnt_out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < m;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t v = static_cast<int64_t>(vals[i]);
    int64_t klo = (v << 32);                       // (v, 0x00000000)
    int64_t khi = (v << 32) | 0xFFFFFFFFLL;        // (v, 0xFFFFFFFF)
    int64_t lo = lower_bound_i64(key12, n, klo);
    int64_t hi = upper_bound_i64(key12, n, khi);
    lo_out[i] = lo;
    cnt_out[i] = static_cast<int32_t>(hi - lo);
  }
}

// ---- merge-path variant for SORTED probes ---------------------------------
// When the probe keys are themselves sorted (PSO-slice star/merge chains),
// a fully-parallel boundary pass computes each 256-probe tile's index
// window; the count pass then searches only inside its tile window, which
// is small and cache-hot — the probe becomes a merge join.
constexpr int kTile = 768;

__global__ void tile_bounds(const int64_t* __restrict__ key12, int64_t n,
                            const int64_t* __restrict__ keys, int64_t m,
                            int64_t n_tiles,
                            int64_t* __restrict__ win_lo,
                            int64_t* __restrict__ win_hi) {
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; t < n_tiles;
       t += (int64_t)gridDim.x * blockDim.x) {
    int64_t first = keys[t * kTile];
    int64_t last = keys[min((t + 1) * kTile - 1, m - 1)];
    win_lo[t] = lower_bound_i64(key12, n, first);
    win_hi[t] = upper_bound_i64(key12, n, last);
  }
}

__global__ void probe_count_exact_sorted(const int64_t* __restrict__ key12,
                                         const int64_t* __restrict__ keys,
                                         int64_t m,
                                         const int64_t* __restrict__ win_lo,
                                         const int64_t* __restrict__ win_hi,
                                         int64_t* __restrict__ lo_out,
                                         int32_t* __restrict__ cnt_out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < m;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t t = i / kTile;
    int64_t wlo = win_lo[t];
    int64_t wspan = win_hi[t] - wlo;
    int64_t k = keys[i];
    int64_t lo = wlo + lower_bound_i64(key12 + wlo, wspan, k);
    int64_t hi = wlo + upper_bound_i64(key12 + wlo, wspan, k);
    lo_out[i] = lo;
    cnt_out[i] = static_cast<int32_t>(hi - lo);
  }
}

// emit: each probe row copies its [lo, lo+cnt) range; writes probe-row index,
// the second packed component (b) and the trailing column (z).
__global__ void probe_emit(const int64_t* __restrict__ key12,
                           const int32_t* __restrict__ z, int64_t m,
                           const int64_t* __restrict__ lo,
                           const int32_t* __restrict__ cnt,
                           const int64_t* __restrict__ offs,
                           int64_t* __restrict__ li_out,
                           int32_t* __restrict__ b_out,
                           int32_t* __restrict__ z_out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < m;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t base = offs[i];
    int64_t l = lo[i];
    int32_t c = cnt[i];
    for (int32_t j = 0; j < c; ++j) {
      li_out[base + j] = i;
      b_out[base + j] = static_cast<int32_t>(key12[l + j] & 0xFFFFFFFFLL);
      z_out[base + j] = z[l + j];
    }
  }
}

// balanced emit: parallel over OUTPUT elements (skew-proof — a probe row
// with 10^4 matches no longer serializes one thread); each output finds its
// probe row by binary search over the offset array.
__global__ void probe_emit_balanced(const int64_t* __restrict__ key12,
                                    const int32_t* __restrict__ z, int64_t m,
                                    const int64_t* __restrict__ lo,
                                    const int64_t* __restrict__ offs,  // [m+1]
                                    int64_t total,
                                    int64_t* __restrict__ li_out,
                                    int32_t* __restrict__ b_out,
                                    int32_t* __restrict__ z_out) {
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    // find probe row i with offs[i] <= idx < offs[i+1]
    int64_t a = 0, b = m;
    while (a + 1 < b) {
      int64_t mid = (a + b) >> 1;
      if (offs[mid] <= idx) a = mid; else b = mid;
    }
    int64_t j = idx - offs[a];
    int64_t src = lo[a] + j;
    li_out[idx] = a;
    b_out[idx] = static_cast<int32_t>(key12[src] & 0xFFFFFFFFLL);
    z_out[idx] = z[src];
  }
}

// ---- fused probe: inline key packing + carry-column emit ------------------
// Builds probe keys (a<<32)|b inside the kernel (a/b each a per-row column
// or a broadcast constant) and gathers up to 4 "carry" columns by probe-row
// index during the emit pass — the engine's bind-join step then needs no
// separate pack2 pass and no separate gathers.
constexpr int kMaxCarry = 4;

struct CarryCols {
  const int32_t* in[kMaxCarry];
  int32_t* out[kMaxCarry];
  int k;
};

__device__ __forceinline__ int64_t fused_key(const int32_t* a_col, int64_t a_const,
                                             const int32_t* b_col, int64_t b_const,
                                             int64_t i) {
  int64_t a = a_col ? static_cast<int64_t>(a_col[i]) : a_const;
  int64_t b = b_col ? static_cast<int64_t>(b_col[i]) : b_const;
  return (a << 32) | (b & 0xFFFFFFFFLL);
}

__global__ void probe_count_fused(const int64_t* __restrict__ key12, int64_t n,
                                  const int32_t* __restrict__ a_col, int64_t a_const,
                                  const int32_t* __restrict__ b_col, int64_t b_const,
                                  int64_t m,
                                  int64_t* __restrict__ lo_out,
                                  int32_t* __restrict__ cnt_out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < m;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t k = fused_key(a_col, a_const, b_col, b_const, i);
    int64_t lo = lower_bound_i64(key12, n, k);
    int64_t hi = upper_bound_i64(key12, n, k);
    lo_out[i] = lo;
    cnt_out[i] = static_cast<int32_t>(hi - lo);
  }
}

__global__ void tile_bounds_fused(const int64_t* __restrict__ key12, int64_t n,
                                  const int32_t* __restrict__ a_col, int64_t a_const,
                                  const int32_t* __restrict__ b_col, int64_t b_const,
                                  int64_t m, int64_t n_tiles,
                                  int64_t* __restrict__ win_lo,
                                  int64_t* __restrict__ win_hi) {
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; t < n_tiles;
       t += (int64_t)gridDim.x * blockDim.x) {
    int64_t first = fused_key(a_col, a_const, b_col, b_const, t * kTile);
    int64_t last = fused_key(a_col, a_const, b_col, b_const,
                             min((t + 1) * kTile - 1, m - 1));
    win_lo[t] = lower_bound_i64(key12, n, first);
    win_hi[t] = upper_bound_i64(key12, n, last);
  }
}

__global__ void probe_count_fused_sorted(const int64_t* __restrict__ key12,
                                         const int32_t* __restrict__ a_col, int64_t a_const,
                                         const int32_t* __restrict__ b_col, int64_t b_const,
                                         int64_t m,
                                         const int64_t* __restrict__ win_lo,
                                         const int64_t* __restrict__ win_hi,
                                         int64_t* __restrict__ lo_out,
                                         int32_t* __restrict__ cnt_out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < m;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t t = i / kTile;
    int64_t wlo = win_lo[t];
    int64_t wspan = win_hi[t] - wlo;
    int64_t k = fused_key(a_col, a_const, b_col, b_const, i);
    int64_t lo = wlo + lower_bound_i64(key12 + wlo, wspan, k);
    int64_t hi = wlo + upper_bound_i64(key12 + wlo, wspan, k);
    lo_out[i] = lo;
    cnt_out[i] = static_cast<int32_t>(hi - lo);
  }
}

__global__ void probe_emit_carry(const int64_t* __restrict__ key12,
                                 const int32_t* __restrict__ z, int64_t m,
                                 const int64_t* __restrict__ lo,
                                 const int32_t* __restrict__ cnt,
                                 const int64_t* __restrict__ offs,
                                 CarryCols carry,
                                 int64_t* __restrict__ li_out,
                                 int32_t* __restrict__ b_out,
                                 int32_t* __restrict__ z_out,
                                 bool emit_b) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < m;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t base = offs[i];
    int64_t l = lo[i];
    int32_t c = cnt[i];
    int32_t cv[kMaxCarry];
    for (int j = 0; j < carry.k; ++j) cv[j] = carry.in[j][i];
    for (int32_t q = 0; q < c; ++q) {
      li_out[base + q] = i;
      if (emit_b)
        b_out[base + q] = static_cast<int32_t>(key12[l + q] & 0xFFFFFFFFLL);
      z_out[base + q] = z[l + q];
      for (int j = 0; j < carry.k; ++j) carry.out[j][base + q] = cv[j];
    }
  }
}

__global__ void probe_emit_carry_balanced(const int64_t* __restrict__ key12,
                                          const int32_t* __restrict__ z, int64_t m,
                                          const int64_t* __restrict__ lo,
                                          const int64_t* __restrict__ offs,  // [m+1]
                                          int64_t total, CarryCols carry,
                                          int64_t* __restrict__ li_out,
                                          int32_t* __restrict__ b_out,
                                          int32_t* __restrict__ z_out,
                                          bool emit_b) {
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    int64_t a = 0, b = m;
    while (a + 1 < b) {
      int64_t mid = (a + b) >> 1;
      if (offs[mid] <= idx) a = mid; else b = mid;
    }
    int64_t j = idx - offs[a];
    int64_t src = lo[a] + j;
    li_out[idx] = a;
    if (emit_b)
      b_out[idx] = static_cast<int32_t>(key12[src] & 0xFFFFFFFFLL);
    z_out[idx] = z[src];
    for (int c = 0; c < carry.k; ++c) carry.out[c][idx] = carry.in[c][a];
  }
}

// ------------------------------------------------------------- K2: hash join
constexpr int kMaxKeyCols = 4;

struct KeyCols {
  const int32_t* c[kMaxKeyCols];
  int k;
};

__device__ __forceinline__ uint64_t mix_hash(const KeyCols& kc, int64_t row) {
  // splitmix-style combine over up to 4 int32 keys
  uint64_t h = 0x9E3779B97F4A7C15ULL;
  for (int j = 0; j < kc.k; ++j) {
    uint64_t x = static_cast<uint32_t>(kc.c[j][row]);
    x *= 0xBF58476D1CE4E5B9ULL;
    x ^= x >> 27;
    h = (h ^ x) * 0x94D049BB133111EBULL;
  }
  return h ^ (h >> 31);
}

__device__ __forceinline__ bool keys_equal(const KeyCols& a, int64_t ra,
                                           const KeyCols& b, int64_t rb) {
  for (int j = 0; j < a.k; ++j)
    if (a.c[j][ra] != b.c[j][rb]) return false;
  return true;
}

__global__ void hj_build(KeyCols build, int64_t r, uint32_t mask,
                         int32_t* __restrict__ heads,
                         int32_t* __restrict__ next) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < r;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t h = static_cast<uint32_t>(mix_hash(build, i)) & mask;
    // chain push: device-scope atomic (cross-XCD safe, guide G12)
    next[i] = atomicExch(&heads[h], static_cast<int32_t>(i));
  }
}

// ---- LDS-staged variant for SMALL build sides -----------------------------
// Build sides up to 8192 rows fit a per-block chained table entirely in LDS
// (32 KB heads + 32 KB next = 64 KB -> 2 blocks/CU, 8 waves).  Every block
// rebuilds the table from the global build columns (<= 8192 LDS atomics,
// amortized over its share of probes) and the probe chain walk then costs
// LDS latency instead of L2 round trips.  This is the selective-query /
// broadcast-join shape: tiny build side, large probe side.
constexpr int kHjLdsSlots = 8192;   // heads, power of 2
constexpr int kHjLdsRows = 8192;    // max build rows

__device__ __forceinline__ void hj_lds_build(KeyCols build, int64_t r,
                                             int32_t* heads, int32_t* next) {
  for (int j = threadIdx.x; j < kHjLdsSlots; j += blockDim.x) heads[j] = -1;
  __syncthreads();
  for (int64_t i = threadIdx.x; i < r; i += blockDim.x) {
    uint32_t h = static_cast<uint32_t>(mix_hash(build, i)) & (kHjLdsSlots - 1);
    next[i] = atomicExch(&heads[h], static_cast<int32_t>(i));
  }
  __syncthreads();
}

__global__ void hj_count_lds(KeyCols probe, int64_t l, KeyCols build,
                             int64_t r, int32_t* __restrict__ cnt) {
  __shared__ int32_t heads[kHjLdsSlots];
  __shared__ int32_t next[kHjLdsRows];
  hj_lds_build(build, r, heads, next);
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < l;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t h = static_cast<uint32_t>(mix_hash(probe, i)) & (kHjLdsSlots - 1);
    int32_t c = 0;
    for (int32_t b = heads[h]; b >= 0; b = next[b])
      if (keys_equal(probe, i, build, b)) ++c;
    cnt[i] = c;
  }
}

__global__ void hj_emit_lds(KeyCols probe, int64_t l, KeyCols build,
                            int64_t r, const int64_t* __restrict__ offs,
                            int64_t* __restrict__ li_out,
                            int64_t* __restrict__ ri_out) {
  __shared__ int32_t heads[kHjLdsSlots];
  __shared__ int32_t next[kHjLdsRows];
  hj_lds_build(build, r, heads, next);
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < l;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t h = static_cast<uint32_t>(mix_hash(probe, i)) & (kHjLdsSlots - 1);
    int64_t base = offs[i];
    for (int32_t b = heads[h]; b >= 0; b = next[b]) {
      if (keys_equal(probe, i, build, b)) {
        li_out[base] = i;
        ri_out[base] = b;
        ++base;
      }
    }
  }
}

__global__ void hj_count(KeyCols probe, int64_t l, KeyCols build,
                         const int32_t* __restrict__ heads,
                         const int32_t* __restrict__ next, uint32_t mask,
                         int32_t* __restrict__ cnt) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < l;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t h = static_cast<uint32_t>(mix_hash(probe, i)) & mask;
    int32_t c = 0;
    for (int32_t r = heads[h]; r >= 0; r = next[r])
      if (keys_equal(probe, i, build, r)) ++c;
    cnt[i] = c;
  }
}

__global__ void hj_emit(KeyCols probe, int64_t l, KeyCols build,
                        const int32_t* __restrict__ heads,
                        const int32_t* __restrict__ next, uint32_t mask,
                        const int64_t* __restrict__ offs,
                        int64_t* __restrict__ li_out,
                        int64_t* __restrict__ ri_out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < l;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t h = static_cast<uint32_t>(mix_hash(probe, i)) & mask;
    int64_t base = offs[i];
    for (int32_t r = heads[h]; r >= 0; r = next[r]) {
      if (keys_equal(probe, i, build, r)) {
        li_out[base] = i;
        ri_out[base] = r;
        ++base;
      }
    }
  }
}

// ------------------------------------------------------------ K5: filter VM
// Postfix bytecode over binding columns.  Opcodes (engine/filter_bytecode.py
// must match):
enum FilterOp : int32_t {
  OP_PUSH_ID = 0,     // arg: column index      -> id stack
  OP_PUSH_CONST_ID,   // arg: literal id        -> id stack
  OP_PUSH_VAL,        // arg: column index      -> val stack (value_col[id])
  OP_PUSH_CONST_VAL,  // arg: f64 const index   -> val stack
  OP_EQ_ID,           // id x id -> bool
  OP_NE_ID,
  OP_LT, OP_GT, OP_LE, OP_GE,    // val x val -> bool
  OP_EQ_VAL, OP_NE_VAL,
  OP_ADD, OP_SUB, OP_MUL, OP_DIV,  // val x val -> val
  OP_AND, OP_OR, OP_NOT,           // bool
  OP_BOUND,           // arg: column index -> bool (id != UNBOUND)
  OP_IS_TRIPLE,       // id -> bool
  OP_PUSH_TRUE, OP_PUSH_FALSE,
};

constexpr int kMaxStack = 16;
constexpr int kMaxCols = 16;
constexpr int32_t kUnbound = -1;

struct FilterProg {
  const int32_t* ops;      // [n_ops]
  const int32_t* args;     // [n_ops]
  const double* consts;    // f64 constant pool
  int n_ops;
};

struct BindCols {
  const int32_t* c[kMaxCols];
  int k;
};

__global__ void filter_eval(FilterProg prog, BindCols cols,
                            const double* __restrict__ value_col,
                            int64_t value_n, int64_t n,
                            bool* __restrict__ out) {
  for (int64_t row = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; row < n;
       row += (int64_t)gridDim.x * blockDim.x) {
    int32_t ids[kMaxStack];
    double vals[kMaxStack];
    bool bools[kMaxStack];
    bool ok[kMaxStack];   // NaN / div0 validity per val slot
    int si = 0, sv = 0, sb = 0;
    for (int pc = 0; pc < prog.n_ops; ++pc) {
      int32_t op = prog.ops[pc];
      int32_t a = prog.args[pc];
      switch (op) {
        case OP_PUSH_ID: ids[si++] = cols.c[a][row]; break;
        case OP_PUSH_CONST_ID: ids[si++] = a; break;
        case OP_PUSH_VAL: {
          int32_t id = cols.c[a][row];
          int64_t u = static_cast<uint32_t>(id);
          double v = (u < value_n && id != kUnbound) ? value_col[u] : 0.0;
          vals[sv] = v; ok[sv] = (id != kUnbound); ++sv;
          break;
        }
        case OP_PUSH_CONST_VAL: vals[sv] = prog.consts[a]; ok[sv] = true; ++sv; break;
        case OP_EQ_ID: { bool r = ids[si-2] == ids[si-1] && ids[si-2] != kUnbound;
                         si -= 2; bools[sb++] = r; break; }
        case OP_NE_ID: { bool r = ids[si-2] != ids[si-1] && ids[si-2] != kUnbound
                                  && ids[si-1] != kUnbound;
                         si -= 2; bools[sb++] = r; break; }
        case OP_LT: case OP_GT: case OP_LE: case OP_GE:
        case OP_EQ_VAL: case OP_NE_VAL: {
          double x = vals[sv-2], y = vals[sv-1];
          bool v = ok[sv-2] && ok[sv-1];
          sv -= 2;
          bool r = false;
          if (op == OP_LT) r = x < y;
          else if (op == OP_GT) r = x > y;
          else if (op == OP_LE) r = x <= y;
          else if (op == OP_GE) r = x >= y;
          else if (op == OP_EQ_VAL) r = x == y;
          else r = x != y;
          bools[sb++] = r && v;
          break;
        }
        case OP_ADD: case OP_SUB: case OP_MUL: case OP_DIV: {
          double x = vals[sv-2], y = vals[sv-1];
          bool v = ok[sv-2] && ok[sv-1];
          sv -= 2;
          double r = 0.0;
          if (op == OP_ADD) r = x + y;
          else if (op == OP_SUB) r = x - y;
          else if (op == OP_MUL) r = x * y;
          else { if (y == 0.0) v = false; else r = x / y; }
          vals[sv] = r; ok[sv] = v; ++sv;
          break;
        }
        case OP_AND: { bool r = bools[sb-2] && bools[sb-1]; sb -= 2; bools[sb++] = r; break; }
        case OP_OR:  { bool r = bools[sb-2] || bools[sb-1]; sb -= 2; bools[sb++] = r; break; }
        case OP_NOT: bools[sb-1] = !bools[sb-1]; break;
        case OP_BOUND: bools[sb++] = cols.c[a][row] != kUnbound; break;
        case OP_IS_TRIPLE: { int32_t id = ids[--si];
                             bools[sb++] = (id < 0) && (id != kUnbound); break; }
        case OP_PUSH_TRUE: bools[sb++] = true; break;
        case OP_PUSH_FALSE: bools[sb++] = false; break;
      }
    }
    out[row] = sb > 0 ? bools[sb-1] : false;
  }
}

inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

}  // namespace

// --------------------------------------------- fused chain-count (K1+K2+K4)
// COUNT(*) over a probe chain whose hop keys all come from the SEED scan:
//   count = sum over seed rows of prod_h |matches_h(key_h(row))|
// One kernel, no intermediate materialization.  Hop regions are
// pre-narrowed to their predicate slice on the host.
constexpr int kMaxHops = 4;

struct ChainHops {
  // narrowed sorted regions, stored as the 32-bit LOW components only
  // (the high word is the hop's constant predicate — dropping it halves
  // the streamed bytes and doubles the LDS window capacity); ascending in
  // UNSIGNED 32-bit order, so all searches compare as uint32
  const int32_t* key32[kMaxHops];
  int64_t n[kMaxHops];
  int32_t src[kMaxHops];           // 0: seed B component, 1: seed Z column
  const unsigned long long* table[kMaxHops];  // optional count table (u64)
  const uint32_t* table32[kMaxHops];  // packed (val<<7)|count u32 variant
  // DIRECT-indexed count table (dense value spaces, e.g. a contiguous
  // id block): counts[v - dmin] — consecutive seed values hit
  // consecutive slots, so the probe loads COALESCE instead of hashing
  // to random lines
  const uint32_t* direct[kMaxHops];
  int64_t dmin[kMaxHops];
  int64_t dlen[kMaxHops];
  int64_t tmask[kMaxHops];
  int k;
  // stage a window through LDS only when it exceeds this row count:
  // small windows (few cache lines, shared by the whole block) search
  // L1/L2 directly without the staging barrier
  int lds_min = 0;
};

__device__ __forceinline__ int64_t lower_bound_u32(
    const int32_t* __restrict__ a, int64_t n, uint32_t key) {
  int64_t lo = 0, hi = n;
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (static_cast<uint32_t>(a[mid]) < key) lo = mid + 1; else hi = mid;
  }
  return lo;
}

__device__ __forceinline__ int64_t upper_bound_u32(
    const int32_t* __restrict__ a, int64_t n, uint32_t key) {
  int64_t lo = 0, hi = n;
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (static_cast<uint32_t>(a[mid]) <= key) lo = mid + 1; else hi = mid;
  }
  return lo;
}

// packed-u32 count table: halves the table footprint (better L2 residency
// beside the seed stream) when every value < 2^25 and count < 2^7
constexpr uint32_t kTbl32Empty = 0xFFFFFFFFu;

// ---- per-hop COUNT tables -------------------------------------------------
// A small hop region (e.g. the 142k-row locatedIn slice under 14.2M seeds)
// makes every seed pay a ~17-level binary search = ~17 distinct L2 lines.
// Instead the host collapses the region to (value, count) pairs once per
// store version and an open-addressing table turns the hop into ONE 8-byte
// L2 load (entry packed (val<<32)|count, empty slot = all-ones).
constexpr unsigned long long kTblEmpty = ~0ull;

__device__ __forceinline__ uint32_t h32(uint32_t x) {
  x *= 2654435761u;
  x ^= x >> 16;
  return x;
}

// ---- K8: database statistics gather ---------------------------------------
// One pass over the committed (s, p, o) columns computes everything the
// cost model needs: per-predicate row counts, global distinct
// subjects/objects, and per-predicate distinct subjects/objects (as
// (p,x) pair claims).  Presence tables are open-addressing claims
// (atomicCAS); row counts aggregate through a per-block LDS histogram
// (predicate sets are tiny); distinct tallies use per-thread
// accumulators folded by one wave reduction (guide G12).
// Ref parity: streamertail_optimizer/stats/database_stats.rs:18-158.
constexpr int kStatsPreds = 1024;  // open-addressing predicate slots
constexpr uint32_t kStatsEmpty = 0xFFFFFFFFu;

__device__ __forceinline__ int stats_pred_slot(uint32_t* pred_keys,
                                               uint32_t pkey) {
  uint32_t slot = h32(pkey) & (kStatsPreds - 1);
  for (;;) {
    uint32_t cur = __hip_atomic_load(&pred_keys[slot], __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_AGENT);
    if (cur == pkey) return static_cast<int>(slot);
    if (cur == kStatsEmpty) {
      uint32_t prev = atomicCAS(&pred_keys[slot], kStatsEmpty, pkey);
      if (prev == kStatsEmpty || prev == pkey)
        return static_cast<int>(slot);
      continue;  // lost the race to a different key: re-inspect the slot
    }
    slot = (slot + 1) & (kStatsPreds - 1);
  }
}

__device__ __forceinline__ bool stats_claim_u32(uint32_t* tbl, uint32_t mask,
                                                uint32_t v) {
  uint32_t slot = h32(v) & mask;
  for (;;) {
    uint32_t cur = __hip_atomic_load(&tbl[slot], __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_AGENT);
    if (cur == v) return false;
    if (cur == kStatsEmpty) {
      uint32_t prev = atomicCAS(&tbl[slot], kStatsEmpty, v);
      if (prev == kStatsEmpty) return true;
      if (prev == v) return false;
      continue;
    }
    slot = (slot + 1) & mask;
  }
}

__device__ __forceinline__ bool stats_claim_u64(unsigned long long* tbl,
                                                uint32_t mask,
                                                unsigned long long v) {
  uint32_t slot = h32(static_cast<uint32_t>(v)
                      ^ static_cast<uint32_t>(v >> 32)) & mask;
  for (;;) {
    unsigned long long cur = __hip_atomic_load(
        &tbl[slot], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    if (cur == v) return false;
    if (cur == ~0ull) {
      unsigned long long prev = atomicCAS(&tbl[slot], ~0ull, v);
      if (prev == ~0ull) return true;
      if (prev == v) return false;
      continue;
    }
    slot = (slot + 1) & mask;
  }
}

__global__ void stats_gather_kernel(
    const int32_t* __restrict__ s, const int32_t* __restrict__ p,
    const int32_t* __restrict__ o, int64_t n, uint32_t* pred_keys,
    unsigned long long* pred_rows, unsigned long long* pred_ds,
    unsigned long long* pred_do, uint32_t* s_tbl, uint32_t* o_tbl,
    uint32_t so_mask, unsigned long long* ps_tbl,
    unsigned long long* po_tbl, uint32_t pair_mask,
    unsigned long long* g_counts /* [0]=distinct_s [1]=distinct_o */) {
  __shared__ uint32_t lds_rows[kStatsPreds];
  for (int i = threadIdx.x; i < kStatsPreds; i += blockDim.x)
    lds_rows[i] = 0;
  __syncthreads();
  unsigned long long acc_s = 0, acc_o = 0;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t pp = static_cast<uint32_t>(p[i]);
    uint32_t ss = static_cast<uint32_t>(s[i]);
    uint32_t oo = static_cast<uint32_t>(o[i]);
    int slot = stats_pred_slot(pred_keys, pp);
    atomicAdd(&lds_rows[slot], 1u);
    if (stats_claim_u32(s_tbl, so_mask, ss)) ++acc_s;
    if (stats_claim_u32(o_tbl, so_mask, oo)) ++acc_o;
    unsigned long long base = static_cast<unsigned long long>(pp) << 32;
    if (stats_claim_u64(ps_tbl, pair_mask, base | ss))
      atomicAdd(&pred_ds[slot], 1ull);
    if (stats_claim_u64(po_tbl, pair_mask, base | oo))
      atomicAdd(&pred_do[slot], 1ull);
  }
  for (int off = 32; off > 0; off >>= 1) {
    acc_s += __shfl_down(acc_s, off);
    acc_o += __shfl_down(acc_o, off);
  }
  if ((threadIdx.x & 63) == 0) {
    if (acc_s) atomicAdd(&g_counts[0], acc_s);
    if (acc_o) atomicAdd(&g_counts[1], acc_o);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < kStatsPreds; i += blockDim.x)
    if (lds_rows[i])
      atomicAdd(&pred_rows[i], static_cast<unsigned long long>(lds_rows[i]));
}

std::vector<at::Tensor> stats_gather(at::Tensor s, at::Tensor p,
                                     at::Tensor o) {
  TORCH_CHECK(s.is_cuda() && p.is_cuda() && o.is_cuda(),
              "stats_gather: device columns required");
  TORCH_CHECK(s.dtype() == at::kInt && p.dtype() == at::kInt
              && o.dtype() == at::kInt);
  int64_t n = s.numel();
  TORCH_CHECK(p.numel() == n && o.numel() == n);
  int64_t so_size = 64;
  while (so_size < 2 * n) so_size <<= 1;
  auto opt_i = s.options();
  auto opt_l = s.options().dtype(at::kLong);
  auto pred_keys = at::full({kStatsPreds}, -1, opt_i);
  auto pred_rows = at::zeros({kStatsPreds}, opt_l);
  auto pred_ds = at::zeros({kStatsPreds}, opt_l);
  auto pred_do = at::zeros({kStatsPreds}, opt_l);
  auto s_tbl = at::full({so_size}, -1, opt_i);
  auto o_tbl = at::full({so_size}, -1, opt_i);
  auto ps_tbl = at::full({so_size}, -1, opt_l);
  auto po_tbl = at::full({so_size}, -1, opt_l);
  auto g_counts = at::zeros({2}, opt_l);
  if (n > 0) {
    hipLaunchKernelGGL(
        stats_gather_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
        cur_stream(), s.data_ptr<int32_t>(), p.data_ptr<int32_t>(),
        o.data_ptr<int32_t>(),
        n, reinterpret_cast<uint32_t*>(pred_keys.data_ptr<int32_t>()),
        reinterpret_cast<unsigned long long*>(pred_rows.data_ptr<int64_t>()),
        reinterpret_cast<unsigned long long*>(pred_ds.data_ptr<int64_t>()),
        reinterpret_cast<unsigned long long*>(pred_do.data_ptr<int64_t>()),
        reinterpret_cast<uint32_t*>(s_tbl.data_ptr<int32_t>()),
        reinterpret_cast<uint32_t*>(o_tbl.data_ptr<int32_t>()),
        static_cast<uint32_t>(so_size - 1),
        reinterpret_cast<unsigned long long*>(ps_tbl.data_ptr<int64_t>()),
        reinterpret_cast<unsigned long long*>(po_tbl.data_ptr<int64_t>()),
        static_cast<uint32_t>(so_size - 1),
        reinterpret_cast<unsigned long long*>(g_counts.data_ptr<int64_t>()));
    HIP_OK(hipGetLastError());
  }
  return {pred_keys, pred_rows, pred_ds, pred_do, g_counts};
}


__global__ void count_table_insert(const int64_t* __restrict__ entries,
                                   int64_t u,
                                   unsigned long long* __restrict__ table,
                                   uint32_t mask) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < u;
       i += (int64_t)gridDim.x * blockDim.x) {
    unsigned long long e = static_cast<unsigned long long>(entries[i]);
    uint32_t slot = h32(static_cast<uint32_t>(e >> 32)) & mask;
    while (atomicCAS(&table[slot], kTblEmpty, e) != kTblEmpty)
      slot = (slot + 1) & mask;
  }
}

__global__ void count_table_insert32(const int32_t* __restrict__ entries,
                                     int64_t u,
                                     uint32_t* __restrict__ table,
                                     uint32_t mask) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < u;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t e = static_cast<uint32_t>(entries[i]);
    uint32_t slot = h32(e >> 7) & mask;
    while (atomicCAS(&table[slot], kTbl32Empty, e) != kTbl32Empty)
      slot = (slot + 1) & mask;
  }
}

at::Tensor build_count_table32(at::Tensor packed_entries) {
  TORCH_CHECK(packed_entries.is_cuda()
              && packed_entries.dtype() == at::kInt);
  int64_t u = packed_entries.numel();
  int64_t size = 64;
  while (size < 2 * u) size <<= 1;
  auto table = at::full({size}, -1, packed_entries.options());
  if (u > 0) {
    hipLaunchKernelGGL(count_table_insert32, dim3(grid_for(u)), dim3(kBlock),
                       0, cur_stream(), packed_entries.data_ptr<int32_t>(),
                       u,
                       reinterpret_cast<uint32_t*>(
                           table.data_ptr<int32_t>()),
                       static_cast<uint32_t>(size - 1));
    HIP_OK(hipGetLastError());
  }
  return table;
}

at::Tensor build_count_table(at::Tensor packed_entries) {
  TORCH_CHECK(packed_entries.is_cuda()
              && packed_entries.dtype() == at::kLong);
  int64_t u = packed_entries.numel();
  int64_t size = 64;
  while (size < 2 * u) size <<= 1;
  auto table = at::full({size}, -1, packed_entries.options());
  if (u > 0) {
    hipLaunchKernelGGL(count_table_insert, dim3(grid_for(u)), dim3(kBlock), 0,
                       cur_stream(), packed_entries.data_ptr<int64_t>(), u,
                       reinterpret_cast<unsigned long long*>(
                           table.data_ptr<int64_t>()),
                       static_cast<uint32_t>(size - 1));
    HIP_OK(hipGetLastError());
  }
  return table;
}

// per-(tile, src0-hop) windows: the seed region is subject-sorted, so the
// B component is monotone — each 256-seed tile's src-0 hop keys fall in a
// narrow window of that hop's region (merge-path; computed in parallel).
__global__ void chain_tile_bounds(const int32_t* __restrict__ seed_b,
                                  int64_t m, int64_t n_tiles, ChainHops hops,
                                  int64_t* __restrict__ win,  // [n_tiles][k][2]
                                  unsigned long long* __restrict__ total_zero
                                  ) {
  if (total_zero && blockIdx.x == 0 && threadIdx.x == 0)
    *total_zero = 0ULL;  // saves the serve path a hipMemsetAsync enqueue
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; t < n_tiles;
       t += (int64_t)gridDim.x * blockDim.x) {
    uint32_t first_b = static_cast<uint32_t>(seed_b[t * kTile]);
    uint32_t last_b =
        static_cast<uint32_t>(seed_b[min((t + 1) * kTile - 1, m - 1)]);
    for (int h = 0; h < hops.k; ++h) {
      int64_t lo = 0, hi = hops.n[h];
      if (hops.table[h] != nullptr || hops.table32[h] != nullptr
          || hops.direct[h] != nullptr) {
        lo = hi = 0;  // hashed/direct hop: window unused
      } else if (hops.src[h] == 0) {
        lo = lower_bound_u32(hops.key32[h], hops.n[h], first_b);
        hi = upper_bound_u32(hops.key32[h], hops.n[h], last_b);
      }
      win[(t * hops.k + h) * 2] = lo;
      win[(t * hops.k + h) * 2 + 1] = hi;
    }
  }
}

// LDS staging: a tile's src-0 window (avg ~|region|/n_tiles rows) is probed
// by all 256 threads of the block — stage it through LDS once and search
// there instead of hammering the same L1/L2 lines (guide: LDS-staged
// probe structures).  16 KB leaves occupancy at 10 blocks/CU.
constexpr int kChainLds = 4096;   // int32 rows: 16 KB, 10 blocks/CU

// each thread owns kSub seeds of its tile (kTile = kSub * kBlock): the
// independent probe chains per thread double the memory-level
// parallelism, and the per-tile costs (window bounds, LDS staging
// barriers) amortize over twice the seeds
constexpr int kSub = kTile / kBlock;

__global__ void chain_count_kernel(const int32_t* __restrict__ seed_b,
                                   const int32_t* __restrict__ seed_z,
                                   int64_t m, ChainHops hops,
                                   const int64_t* __restrict__ win,
                                   unsigned long long* __restrict__ total) {
  __shared__ int32_t lds[kChainLds];
  unsigned long long acc = 0;
  // block-uniform tile iteration (every thread of the block is in the same
  // tile) so the cooperative LDS loads can barrier safely
  for (int64_t t = blockIdx.x; t * kTile < m; t += gridDim.x) {
    uint32_t b_c[kSub], z_c[kSub];
    bool act[kSub];
    unsigned long long prod[kSub];
    for (int u = 0; u < kSub; ++u) {
      int64_t i = t * kTile + u * kBlock + threadIdx.x;
      act[u] = i < m;
      b_c[u] = act[u] ? static_cast<uint32_t>(seed_b[i]) : 0;
      z_c[u] = act[u] ? static_cast<uint32_t>(seed_z[i]) : 0;
      prod[u] = 1;
    }
    // no early exit: the hop searches are independent dependent-load
    // chains — letting them all issue gives the scheduler ILP to hide
    // L2 latency
    for (int h = 0; h < hops.k; ++h) {
      if (hops.direct[h] != nullptr) {
        // dense-value hop: counts[v - dmin], coalesced across the wave
        for (int u = 0; u < kSub; ++u) {
          if (!act[u]) continue;
          uint32_t v = hops.src[h] == 0 ? b_c[u] : z_c[u];
          uint64_t off = static_cast<uint64_t>(
              static_cast<int64_t>(v) - hops.dmin[h]);
          prod[u] *= off < static_cast<uint64_t>(hops.dlen[h])
                         ? hops.direct[h][off] : 0u;
        }
        continue;
      }
      if (hops.table32[h] != nullptr) {
        // packed count-table hop: one 4-byte L2 load per seed
        uint32_t mask = static_cast<uint32_t>(hops.tmask[h]);
        for (int u = 0; u < kSub; ++u) {
          if (!act[u]) continue;
          uint32_t v = hops.src[h] == 0 ? b_c[u] : z_c[u];
          uint32_t slot = h32(v) & mask;
          unsigned long long cnt = 0;
          for (;;) {
            uint32_t e = hops.table32[h][slot];
            if (e == kTbl32Empty) break;
            if ((e >> 7) == v) {
              cnt = e & 0x7Fu;
              break;
            }
            slot = (slot + 1) & mask;
          }
          prod[u] *= cnt;
        }
        continue;
      }
      if (hops.table[h] != nullptr) {
        // count-table hop: one 8-byte L2 load (vs log2(n) lines)
        uint32_t mask = static_cast<uint32_t>(hops.tmask[h]);
        for (int u = 0; u < kSub; ++u) {
          if (!act[u]) continue;
          uint32_t v = hops.src[h] == 0 ? b_c[u] : z_c[u];
          uint32_t slot = h32(v) & mask;
          unsigned long long cnt = 0;
          for (;;) {
            unsigned long long e = hops.table[h][slot];
            if (e == kTblEmpty) break;
            if (static_cast<uint32_t>(e >> 32) == v) {
              cnt = e & 0xFFFFFFFFull;
              break;
            }
            slot = (slot + 1) & mask;
          }
          prod[u] *= cnt;
        }
        continue;
      }
      int64_t wlo = win[(t * hops.k + h) * 2];
      int64_t wspan = win[(t * hops.k + h) * 2 + 1] - wlo;
      const int32_t* base = hops.key32[h] + wlo;
      if (hops.src[h] == 0 && wspan <= kChainLds
          && wspan > hops.lds_min) {
        // cooperative stage + barrier: search LDS, not HBM/L2
        for (int64_t j = threadIdx.x; j < wspan; j += blockDim.x)
          lds[j] = base[j];
        __syncthreads();
        for (int u = 0; u < kSub; ++u) {
          if (!act[u]) continue;
          uint32_t key = hops.src[h] == 0 ? b_c[u] : z_c[u];
          int64_t lo = lower_bound_u32(lds, wspan, key);
          int64_t hi = lo;
          while (hi < wspan && hi - lo < 4
                 && static_cast<uint32_t>(lds[hi]) == key) ++hi;
          if (hi - lo == 4 && hi < wspan
              && static_cast<uint32_t>(lds[hi]) == key)
            hi = lo + upper_bound_u32(lds + lo, wspan - lo, key);
          prod[u] *= static_cast<unsigned long long>(hi - lo);
        }
        __syncthreads();  // before the next hop reuses the buffer
        continue;
      }
      for (int u = 0; u < kSub; ++u) {
        if (!act[u]) continue;
        uint32_t key = hops.src[h] == 0 ? b_c[u] : z_c[u];
        int64_t lo = lower_bound_u32(base, wspan, key);
        // match runs are tiny (one object per subject in typical star
        // data): walk forward a few cache-hot slots instead of paying a
        // second full log2(wspan) dependent-load chain
        int64_t hi = lo;
        while (hi < wspan && hi - lo < 4
               && static_cast<uint32_t>(base[hi]) == key) ++hi;
        if (hi - lo == 4 && hi < wspan
            && static_cast<uint32_t>(base[hi]) == key)
          hi = lo + upper_bound_u32(base + lo, wspan - lo, key);
        prod[u] *= static_cast<unsigned long long>(hi - lo);
      }
    }
    for (int u = 0; u < kSub; ++u)
      if (act[u]) acc += prod[u];
  }
  // wave reduction then one device-scope atomic per wave (guide G12)
  for (int off = 32; off > 0; off >>= 1)
    acc += __shfl_down(acc, off);
  if ((threadIdx.x & 63) == 0 && acc)
    atomicAdd(total, acc);
}

// Capture-friendly variant: caller owns the window + total buffers, no
// allocation and no device->host sync inside — the launch pair can be
// recorded into a hipGraph and replayed per query (launch-bound path).
void chain_count_into(at::Tensor seed_b, at::Tensor seed_z,
                      std::vector<at::Tensor> hop_key32,
                      std::vector<int64_t> hop_src,
                      std::vector<at::Tensor> hop_table,
                      std::vector<int64_t> hop_dmin,
                      at::Tensor win, at::Tensor total) {
  TORCH_CHECK(seed_b.is_cuda() && seed_b.dtype() == at::kInt
              && seed_z.is_cuda());
  TORCH_CHECK(win.dtype() == at::kLong && total.dtype() == at::kLong);
  int64_t m = seed_b.numel();
  ChainHops hops{};
  if (const char* e = getenv("KOLIBRIE_CHAIN_LDS_MIN"))
    hops.lds_min = atoi(e);
  hops.k = static_cast<int>(hop_key32.size());
  for (size_t h = 0; h < hop_key32.size(); ++h) {
    TORCH_CHECK(hop_key32[h].dtype() == at::kInt);
    hops.key32[h] = hop_key32[h].data_ptr<int32_t>();
    hops.n[h] = hop_key32[h].numel();
    hops.src[h] = static_cast<int32_t>(hop_src[h]);
    if (hop_table[h].numel() > 0) {
      bool is_direct = h < hop_dmin.size() && hop_dmin[h] >= 0;
      if (is_direct) {
        hops.direct[h] = reinterpret_cast<const uint32_t*>(
            hop_table[h].data_ptr<int32_t>());
        hops.dmin[h] = hop_dmin[h];
        hops.dlen[h] = hop_table[h].numel();
      } else if (hop_table[h].dtype() == at::kInt) {
        hops.table32[h] = reinterpret_cast<const uint32_t*>(
            hop_table[h].data_ptr<int32_t>());
      } else {
        hops.table[h] = reinterpret_cast<const unsigned long long*>(
            hop_table[h].data_ptr<int64_t>());
      }
      hops.tmask[h] = hop_table[h].numel() - 1;
    }
  }
  auto stream = cur_stream();
  total.zero_();
  if (m == 0) return;
  int64_t n_tiles = (m + kTile - 1) / kTile;
  TORCH_CHECK(win.numel() >= n_tiles * hops.k * 2, "win buffer too small");
  hipLaunchKernelGGL(chain_tile_bounds, dim3(grid_for(n_tiles)),
                     dim3(kBlock), 0, stream,
                     seed_b.data_ptr<int32_t>(), m, n_tiles, hops,
                     win.data_ptr<int64_t>(),
                     static_cast<unsigned long long*>(nullptr));
  HIP_OK(hipGetLastError());
  hipLaunchKernelGGL(chain_count_kernel, dim3(grid_for(m)), dim3(kBlock), 0,
                     stream, seed_b.data_ptr<int32_t>(),
                     seed_z.data_ptr<int32_t>(), m, hops,
                     win.data_ptr<int64_t>(),
                     reinterpret_cast<unsigned long long*>(
                         total.data_ptr<int64_t>()));
  HIP_OK(hipGetLastError());
}

// ---- C++ serving path (crush the per-query host
// floor).  All launch arguments for a cached COUNT chain live in a
// C++-side registry; one pybind call then does: zero the accumulator,
// two direct kernel launches, an 8-byte D2H into pinned memory and a
// stream sync — no Python plan walk, no torch dispatcher, no graph
// replay floor (direct launches are ~3-4 µs each vs 10-16 µs replay,
// MI355X guide §graph-replay-floor).
struct ChainServe {
  at::Tensor seed_b, seed_z, win, total;
  std::vector<at::Tensor> keep;
  ChainHops hops{};
  int64_t m = 0, n_tiles = 0;
  int64_t* pinned = nullptr;
  ~ChainServe() {
    if (pinned) (void)hipHostFree(pinned);
  }
};
static std::vector<std::unique_ptr<ChainServe>> g_chain_serves;

int64_t register_chain_serve(at::Tensor seed_b, at::Tensor seed_z,
                             std::vector<at::Tensor> hop_key32,
                             std::vector<int64_t> hop_src,
                             std::vector<at::Tensor> hop_table,
                             std::vector<int64_t> hop_dmin) {
  TORCH_CHECK(seed_b.is_cuda() && seed_b.dtype() == at::kInt
              && seed_z.is_cuda());
  auto cs = std::make_unique<ChainServe>();
  cs->seed_b = seed_b;
  cs->seed_z = seed_z;
  cs->m = seed_b.numel();
  if (const char* e = getenv("KOLIBRIE_CHAIN_LDS_MIN"))
    cs->hops.lds_min = atoi(e);
  cs->hops.k = static_cast<int>(hop_key32.size());
  for (size_t h = 0; h < hop_key32.size(); ++h) {
    TORCH_CHECK(hop_key32[h].dtype() == at::kInt);
    cs->keep.push_back(hop_key32[h]);
    cs->keep.push_back(hop_table[h]);
    cs->hops.key32[h] = hop_key32[h].data_ptr<int32_t>();
    cs->hops.n[h] = hop_key32[h].numel();
    cs->hops.src[h] = static_cast<int32_t>(hop_src[h]);
    if (hop_table[h].numel() > 0) {
      bool is_direct = h < hop_dmin.size() && hop_dmin[h] >= 0;
      if (is_direct) {
        cs->hops.direct[h] = reinterpret_cast<const uint32_t*>(
            hop_table[h].data_ptr<int32_t>());
        cs->hops.dmin[h] = hop_dmin[h];
        cs->hops.dlen[h] = hop_table[h].numel();
      } else if (hop_table[h].dtype() == at::kInt) {
        cs->hops.table32[h] = reinterpret_cast<const uint32_t*>(
            hop_table[h].data_ptr<int32_t>());
      } else {
        cs->hops.table[h] = reinterpret_cast<const unsigned long long*>(
            hop_table[h].data_ptr<int64_t>());
      }
      cs->hops.tmask[h] = hop_table[h].numel() - 1;
    }
  }
  cs->n_tiles = (cs->m + kTile - 1) / kTile;
  cs->win = at::empty({std::max<int64_t>(1, cs->n_tiles * cs->hops.k * 2)},
                      seed_b.options().dtype(at::kLong));
  // total[0] = accumulator, total[1] = constant 1 (the completion flag's
  // device source: same-stream copies are ordered, so once the flag
  // lands in pinned memory the count before it is valid)
  cs->total = at::zeros({2}, seed_b.options().dtype(at::kLong));
  cs->total[1] = 1;
  HIP_OK(hipHostMalloc(reinterpret_cast<void**>(&cs->pinned), 16));
  // the per-tile hop windows depend only on the CACHED seeds + hop
  // regions (immutable for this registration's lifetime: a store-version
  // bump re-registers), so compute them ONCE here instead of per query —
  // chain_tile_bounds was ~8% of the serve step.  It also zeroes the
  // accumulator, so the first serve call starts clean.
  if (cs->m > 0) {
    hipLaunchKernelGGL(chain_tile_bounds, dim3(grid_for(cs->n_tiles)),
                       dim3(kBlock), 0, cur_stream(),
                       cs->seed_b.data_ptr<int32_t>(), cs->m, cs->n_tiles,
                       cs->hops, cs->win.data_ptr<int64_t>(),
                       reinterpret_cast<unsigned long long*>(
                           cs->total.data_ptr<int64_t>()));
    HIP_OK(hipGetLastError());
  }
  g_chain_serves.push_back(std::move(cs));
  return static_cast<int64_t>(g_chain_serves.size() - 1);
}

int64_t serve_chain_count(int64_t id) {
  auto& cs = *g_chain_serves.at(id);
  if (cs.m == 0) return 0;
  auto stream = cur_stream();
  int64_t* total_ptr = cs.total.data_ptr<int64_t>();
  volatile int64_t* flag = cs.pinned + 1;
  *flag = 0;
  // windows were precomputed at registration; the accumulator was zeroed
  // there (first call) or by the post-readback memset of the previous call
  hipLaunchKernelGGL(chain_count_kernel, dim3(grid_for(cs.m)), dim3(kBlock),
                     0, stream, cs.seed_b.data_ptr<int32_t>(),
                     cs.seed_z.data_ptr<int32_t>(), cs.m, cs.hops,
                     cs.win.data_ptr<int64_t>(),
                     reinterpret_cast<unsigned long long*>(total_ptr));
  HIP_OK(hipMemcpyAsync(cs.pinned, total_ptr, 16, hipMemcpyDeviceToHost,
                        stream));
  // re-zero for the NEXT call after the readback copy — off the
  // critical path of this query's spin-wait
  HIP_OK(hipMemsetAsync(total_ptr, 0, 8, stream));
  // spin on the pinned flag: hipStreamSynchronize pays tens of µs of
  // scheduler yield latency, the DMA lands in ~µs.  Bounded: fall back
  // to a real sync after ~2e9 spins (GPU hung elsewhere).
  for (int64_t spins = 0; *flag == 0; ++spins) {
#if defined(__x86_64__)
    __builtin_ia32_pause();
#endif
    if (spins > 2000000000LL) {
      HIP_OK(hipStreamSynchronize(stream));
      break;
    }
  }
  return cs.pinned[0];
}

void release_chain_serve(int64_t id) {
  if (id >= 0 && id < static_cast<int64_t>(g_chain_serves.size()))
    g_chain_serves[id].reset();
}

int64_t chain_count(at::Tensor seed_b, at::Tensor seed_z,
                    std::vector<at::Tensor> hop_key32,
                    std::vector<int64_t> hop_src,
                    std::vector<at::Tensor> hop_table,
                    std::vector<int64_t> hop_dmin) {
  TORCH_CHECK(seed_b.is_cuda() && seed_b.dtype() == at::kInt
              && seed_z.is_cuda());
  TORCH_CHECK(hop_key32.size() <= static_cast<size_t>(kMaxHops));
  TORCH_CHECK(hop_key32.size() == hop_src.size()
              && hop_key32.size() == hop_table.size());
  int64_t m = seed_b.numel();
  ChainHops hops{};
  if (const char* e = getenv("KOLIBRIE_CHAIN_LDS_MIN"))
    hops.lds_min = atoi(e);
  hops.k = static_cast<int>(hop_key32.size());
  for (size_t h = 0; h < hop_key32.size(); ++h) {
    TORCH_CHECK(hop_key32[h].is_cuda() && hop_key32[h].dtype() == at::kInt);
    hops.key32[h] = hop_key32[h].data_ptr<int32_t>();
    hops.n[h] = hop_key32[h].numel();
    hops.src[h] = static_cast<int32_t>(hop_src[h]);
    if (hop_table[h].numel() > 0) {
      TORCH_CHECK(hop_table[h].is_cuda()
                  && (hop_table[h].dtype() == at::kLong
                      || hop_table[h].dtype() == at::kInt));
      bool is_direct = h < hop_dmin.size() && hop_dmin[h] >= 0;
      if (is_direct) {
        hops.direct[h] = reinterpret_cast<const uint32_t*>(
            hop_table[h].data_ptr<int32_t>());
        hops.dmin[h] = hop_dmin[h];
        hops.dlen[h] = hop_table[h].numel();
      } else if (hop_table[h].dtype() == at::kInt) {
        hops.table32[h] = reinterpret_cast<const uint32_t*>(
            hop_table[h].data_ptr<int32_t>());
      } else {
        hops.table[h] = reinterpret_cast<const unsigned long long*>(
            hop_table[h].data_ptr<int64_t>());
      }
      hops.tmask[h] = hop_table[h].numel() - 1;
    }

  }
  auto total = at::zeros({1}, seed_b.options().dtype(at::kLong));
  if (m > 0) {
    auto stream = cur_stream();
    int64_t n_tiles = (m + kTile - 1) / kTile;
    auto win = at::empty({n_tiles * hops.k * 2},
                         seed_b.options().dtype(at::kLong));
    hipLaunchKernelGGL(chain_tile_bounds, dim3(grid_for(n_tiles)),
                       dim3(kBlock), 0, stream,
                       seed_b.data_ptr<int32_t>(), m, n_tiles, hops,
                       win.data_ptr<int64_t>(),
                       static_cast<unsigned long long*>(nullptr));
    HIP_OK(hipGetLastError());
    hipLaunchKernelGGL(chain_count_kernel, dim3(grid_for(m)), dim3(kBlock), 0,
                       stream, seed_b.data_ptr<int32_t>(),
                       seed_z.data_ptr<int32_t>(), m, hops,
                       win.data_ptr<int64_t>(),
                       reinterpret_cast<unsigned long long*>(
                           total.data_ptr<int64_t>()));
    HIP_OK(hipGetLastError());
  }
  return total.item<int64_t>();
}

// ------------------------------------------------------------ host launchers

// shared emit phase for the K1 probes: per-probe-row emit for low fanout,
// output-parallel balanced emit for skewed ranges.
static std::vector<at::Tensor> emit_phase(at::Tensor key12, at::Tensor z,
                                          at::Tensor lo, at::Tensor cnt,
                                          int64_t m, hipStream_t stream) {
  auto opts_long = key12.options();
  auto offs = at::cumsum(cnt, 0, at::kLong);
  int64_t total = m > 0 ? offs[-1].item<int64_t>() : 0;
  auto li = at::empty({total}, opts_long);
  auto b = at::empty({total}, z.options());
  auto zz = at::empty({total}, z.options());
  if (total == 0) return {li, b, zz};
  if (total >= 8 * m) {
    auto offs_full = at::zeros({m + 1}, opts_long);
    offs_full.narrow(0, 1, m).copy_(offs);
    hipLaunchKernelGGL(probe_emit_balanced, dim3(grid_for(total)),
                       dim3(kBlock), 0, stream, key12.data_ptr<int64_t>(),
                       z.data_ptr<int32_t>(), m, lo.data_ptr<int64_t>(),
                       offs_full.data_ptr<int64_t>(), total,
                       li.data_ptr<int64_t>(), b.data_ptr<int32_t>(),
                       zz.data_ptr<int32_t>());
  } else {
    auto offs_excl = offs - cnt.to(at::kLong);
    hipLaunchKernelGGL(probe_emit, dim3(grid_for(m)), dim3(kBlock), 0, stream,
                       key12.data_ptr<int64_t>(), z.data_ptr<int32_t>(), m,
                       lo.data_ptr<int64_t>(), cnt.data_ptr<int32_t>(),
                       offs_excl.data_ptr<int64_t>(), li.data_ptr<int64_t>(),
                       b.data_ptr<int32_t>(), zz.data_ptr<int32_t>());
  }
  HIP_OK(hipGetLastError());
  return {li, b, zz};
}

// K1 probe — exact (packed 2-col key) mode.
std::vector<at::Tensor> probe_exact(at::Tensor key12, at::Tensor z,
                                    at::Tensor keys) {
  TORCH_CHECK(key12.is_cuda() && z.is_cuda() && keys.is_cuda(),
              "probe_exact: device tensors required");
  TORCH_CHECK(key12.dtype() == at::kLong && keys.dtype() == at::kLong);
  TORCH_CHECK(z.dtype() == at::kInt);
  auto m = keys.numel();
  auto n = key12.numel();
  auto lo = at::empty({m}, keys.options());
  auto cnt = at::empty({m}, keys.options().dtype(at::kInt));
  auto stream = cur_stream();
  if (m > 0) {
    bool sorted = false;
    if (m >= 262144) {
      // one cheap reduction decides the merge-path variant
      sorted = at::all(keys.slice(0, 1, m) >= keys.slice(0, 0, m - 1))
                   .item<bool>();
    }
    if (sorted) {
      int64_t n_tiles = (m + kTile - 1) / kTile;
      auto wlo = at::empty({n_tiles}, keys.options());
      auto whi = at::empty({n_tiles}, keys.options());
      hipLaunchKernelGGL(tile_bounds, dim3(grid_for(n_tiles)), dim3(kBlock), 0,
                         stream, key12.data_ptr<int64_t>(), n,
                         keys.data_ptr<int64_t>(), m, n_tiles,
                         wlo.data_ptr<int64_t>(), whi.data_ptr<int64_t>());
      HIP_OK(hipGetLastError());
      hipLaunchKernelGGL(probe_count_exact_sorted, dim3(grid_for(m)),
                         dim3(kBlock), 0, stream, key12.data_ptr<int64_t>(),
                         keys.data_ptr<int64_t>(), m, wlo.data_ptr<int64_t>(),
                         whi.data_ptr<int64_t>(), lo.data_ptr<int64_t>(),
                         cnt.data_ptr<int32_t>());
      HIP_OK(hipGetLastError());
    } else {
      hipLaunchKernelGGL(probe_count_exact, dim3(grid_for(m)), dim3(kBlock), 0,
                         stream, key12.data_ptr<int64_t>(), n,
                         keys.data_ptr<int64_t>(), m, lo.data_ptr<int64_t>(),
                         cnt.data_ptr<int32_t>());
      HIP_OK(hipGetLastError());
    }
  }
  return emit_phase(key12, z, lo, cnt, m, stream);
}

// K1 probe — range (1-col prefix) mode.
std::vector<at::Tensor> probe_range(at::Tensor key12, at::Tensor z,
                                    at::Tensor vals) {
  TORCH_CHECK(key12.is_cuda() && z.is_cuda() && vals.is_cuda());
  TORCH_CHECK(vals.dtype() == at::kInt);
  auto m = vals.numel();
  auto n = key12.numel();
  auto lo = at::empty({m}, key12.options());
  auto cnt = at::empty({m}, vals.options());
  auto stream = cur_stream();
  if (m > 0) {
    hipLaunchKernelGGL(probe_count_range, dim3(grid_for(m)), dim3(kBlock), 0,
                       stream, key12.data_ptr<int64_t>(), n,
                       vals.data_ptr<int32_t>(), m, lo.data_ptr<int64_t>(),
                       cnt.data_ptr<int32_t>());
    HIP_OK(hipGetLastError());
  }
  return emit_phase(key12, z, lo, cnt, m, stream);
}

// K1 fused probe: inline key packing, optional merge-path for sorted keys,
// carry columns gathered during emit.  a/b: per-row column or constant.
std::vector<at::Tensor> probe_fused(at::Tensor key12, at::Tensor z,
                                    c10::optional<at::Tensor> a_col,
                                    int64_t a_const,
                                    c10::optional<at::Tensor> b_col,
                                    int64_t b_const,
                                    std::vector<at::Tensor> carry_cols,
                                    bool emit_b) {
  TORCH_CHECK(key12.is_cuda() && z.is_cuda());
  TORCH_CHECK(carry_cols.size() <= static_cast<size_t>(kMaxCarry));
  int64_t m = a_col.has_value() ? a_col->numel()
                                : (b_col.has_value() ? b_col->numel() : 0);
  TORCH_CHECK(m > 0, "probe_fused: need at least one probe column");
  auto n = key12.numel();
  auto opts_long = key12.options();
  auto opts_int = z.options();
  const int32_t* a_ptr = a_col.has_value() ? a_col->data_ptr<int32_t>() : nullptr;
  const int32_t* b_ptr = b_col.has_value() ? b_col->data_ptr<int32_t>() : nullptr;
  // constant leading component: narrow every search to that contiguous
  // region ONCE (e.g. one predicate's slice) — the region usually stays
  // cache-resident, cutting the dependent-load chain per probe
  if (!a_ptr && n > 0) {
    auto reg_keys = at::empty({2}, opts_long);
    reg_keys[0] = (a_const << 32);
    reg_keys[1] = (a_const << 32) | 0xFFFFFFFFLL;
    auto lo_t = at::searchsorted(key12, reg_keys[0], false, false);
    auto hi_t = at::searchsorted(key12, reg_keys[1], false, true);
    int64_t roff = lo_t.item<int64_t>();
    int64_t rend = hi_t.item<int64_t>();
    key12 = key12.narrow(0, roff, rend - roff);
    z = z.narrow(0, roff, rend - roff);
    n = rend - roff;
  }
  auto lo = at::empty({m}, opts_long);
  auto cnt = at::empty({m}, opts_int);
  auto stream = cur_stream();
  // merge-path when the single varying column is sorted (signed order is
  // the packed order when values share a sign; negatives skip the check)
  bool sorted = false;
  if (m >= 262144 && (a_ptr == nullptr) != (b_ptr == nullptr)) {
    const auto& col = a_col.has_value() ? *a_col : *b_col;
    bool need_nonneg = b_col.has_value();  // low 32 bits compare unsigned
    auto mono = at::all(col.slice(0, 1, m) >= col.slice(0, 0, m - 1));
    if (need_nonneg) {
      sorted = (mono & (col[0] >= 0)).item<bool>();
    } else {
      sorted = mono.item<bool>();
    }
  }
  if (sorted) {
    int64_t n_tiles = (m + kTile - 1) / kTile;
    auto wlo = at::empty({n_tiles}, opts_long);
    auto whi = at::empty({n_tiles}, opts_long);
    hipLaunchKernelGGL(tile_bounds_fused, dim3(grid_for(n_tiles)),
                       dim3(kBlock), 0, stream, key12.data_ptr<int64_t>(), n,
                       a_ptr, a_const, b_ptr, b_const, m, n_tiles,
                       wlo.data_ptr<int64_t>(), whi.data_ptr<int64_t>());
    HIP_OK(hipGetLastError());
    hipLaunchKernelGGL(probe_count_fused_sorted, dim3(grid_for(m)),
                       dim3(kBlock), 0, stream, key12.data_ptr<int64_t>(),
                       a_ptr, a_const, b_ptr, b_const, m,
                       wlo.data_ptr<int64_t>(), whi.data_ptr<int64_t>(),
                       lo.data_ptr<int64_t>(), cnt.data_ptr<int32_t>());
  } else {
    hipLaunchKernelGGL(probe_count_fused, dim3(grid_for(m)), dim3(kBlock), 0,
                       stream, key12.data_ptr<int64_t>(), n, a_ptr, a_const,
                       b_ptr, b_const, m, lo.data_ptr<int64_t>(),
                       cnt.data_ptr<int32_t>());
  }
  HIP_OK(hipGetLastError());
  auto offs = at::cumsum(cnt, 0, at::kLong);
  int64_t total = offs[-1].item<int64_t>();
  auto li = at::empty({total}, opts_long);
  auto b_out = at::empty({emit_b ? total : 0}, opts_int);
  auto zz = at::empty({total}, opts_int);
  CarryCols carry{};
  carry.k = static_cast<int>(carry_cols.size());
  std::vector<at::Tensor> carried;
  for (size_t j = 0; j < carry_cols.size(); ++j) {
    TORCH_CHECK(carry_cols[j].is_cuda()
                && carry_cols[j].dtype() == at::kInt
                && carry_cols[j].numel() == m);
    carry.in[j] = carry_cols[j].data_ptr<int32_t>();
    carried.push_back(at::empty({total}, opts_int));
    carry.out[j] = carried.back().data_ptr<int32_t>();
  }
  if (total > 0) {
    int32_t* b_ptr_out = emit_b ? b_out.data_ptr<int32_t>() : nullptr;
    if (total >= 8 * m) {
      auto offs_full = at::zeros({m + 1}, opts_long);
      offs_full.narrow(0, 1, m).copy_(offs);
      hipLaunchKernelGGL(probe_emit_carry_balanced, dim3(grid_for(total)),
                         dim3(kBlock), 0, stream, key12.data_ptr<int64_t>(),
                         z.data_ptr<int32_t>(), m, lo.data_ptr<int64_t>(),
                         offs_full.data_ptr<int64_t>(), total, carry,
                         li.data_ptr<int64_t>(), b_ptr_out,
                         zz.data_ptr<int32_t>(), emit_b);
    } else {
      auto offs_excl = offs - cnt.to(at::kLong);
      hipLaunchKernelGGL(probe_emit_carry, dim3(grid_for(m)), dim3(kBlock), 0,
                         stream, key12.data_ptr<int64_t>(),
                         z.data_ptr<int32_t>(), m, lo.data_ptr<int64_t>(),
                         cnt.data_ptr<int32_t>(), offs_excl.data_ptr<int64_t>(),
                         carry, li.data_ptr<int64_t>(), b_ptr_out,
                         zz.data_ptr<int32_t>(), emit_b);
    }
    HIP_OK(hipGetLastError());
  }
  std::vector<at::Tensor> out = {li, b_out, zz};
  for (auto& t : carried) out.push_back(t);
  return out;
}

// K1 count-only variants: COUNT(*) queries skip the emit pass entirely.
at::Tensor probe_exact_counts(at::Tensor key12, at::Tensor keys) {
  TORCH_CHECK(key12.is_cuda() && keys.is_cuda());
  auto m = keys.numel();
  auto n = key12.numel();
  auto lo = at::empty({m}, keys.options());
  auto cnt = at::empty({m}, keys.options().dtype(at::kInt));
  if (m > 0) {
    hipLaunchKernelGGL(probe_count_exact, dim3(grid_for(m)), dim3(kBlock), 0,
                       cur_stream(), key12.data_ptr<int64_t>(), n,
                       keys.data_ptr<int64_t>(), m, lo.data_ptr<int64_t>(),
                       cnt.data_ptr<int32_t>());
    HIP_OK(hipGetLastError());
  }
  return cnt;
}

at::Tensor probe_range_counts(at::Tensor key12, at::Tensor vals) {
  TORCH_CHECK(key12.is_cuda() && vals.is_cuda());
  auto m = vals.numel();
  auto n = key12.numel();
  auto lo = at::empty({m}, key12.options());
  auto cnt = at::empty({m}, vals.options());
  if (m > 0) {
    hipLaunchKernelGGL(probe_count_range, dim3(grid_for(m)), dim3(kBlock), 0,
                       cur_stream(), key12.data_ptr<int64_t>(), n,
                       vals.data_ptr<int32_t>(), m, lo.data_ptr<int64_t>(),
                       cnt.data_ptr<int32_t>());
    HIP_OK(hipGetLastError());
  }
  return cnt;
}

// K2 hash join, count-only: per-probe-row match counts (no emit).
at::Tensor hash_join_counts(std::vector<at::Tensor> left_cols,
                            std::vector<at::Tensor> right_cols) {
  TORCH_CHECK(!left_cols.empty() && left_cols.size() == right_cols.size());
  TORCH_CHECK(left_cols.size() <= kMaxKeyCols);
  int64_t l = left_cols[0].numel();
  int64_t r = right_cols[0].numel();
  auto opts_int = left_cols[0].options().dtype(at::kInt);
  if (l == 0 || r == 0) return at::zeros({l}, opts_int);
  KeyCols probe{}, build{};
  probe.k = build.k = static_cast<int>(left_cols.size());
  for (size_t j = 0; j < left_cols.size(); ++j) {
    probe.c[j] = left_cols[j].data_ptr<int32_t>();
    build.c[j] = right_cols[j].data_ptr<int32_t>();
  }
  auto stream = cur_stream();
  auto cnt = at::empty({l}, opts_int);
  if (r <= kHjLdsRows) {
    hipLaunchKernelGGL(hj_count_lds, dim3(grid_for(l)), dim3(kBlock), 0,
                       stream, probe, l, build, r, cnt.data_ptr<int32_t>());
    HIP_OK(hipGetLastError());
    return cnt;
  }
  uint64_t h = 1;
  while (h < static_cast<uint64_t>(2 * r)) h <<= 1;
  uint32_t mask = static_cast<uint32_t>(h - 1);
  auto heads = at::full({static_cast<int64_t>(h)}, -1, opts_int);
  auto next = at::empty({r}, opts_int);
  hipLaunchKernelGGL(hj_build, dim3(grid_for(r)), dim3(kBlock), 0, stream,
                     build, r, mask, heads.data_ptr<int32_t>(),
                     next.data_ptr<int32_t>());
  HIP_OK(hipGetLastError());
  hipLaunchKernelGGL(hj_count, dim3(grid_for(l)), dim3(kBlock), 0, stream,
                     probe, l, build, heads.data_ptr<int32_t>(),
                     next.data_ptr<int32_t>(), mask, cnt.data_ptr<int32_t>());
  HIP_OK(hipGetLastError());
  return cnt;
}

// K2 hash join: returns (li, ri) index pairs; multiset semantics.
std::vector<at::Tensor> hash_join(std::vector<at::Tensor> left_cols,
                                  std::vector<at::Tensor> right_cols) {
  TORCH_CHECK(!left_cols.empty() && left_cols.size() == right_cols.size());
  TORCH_CHECK(left_cols.size() <= kMaxKeyCols, "hash_join: at most 4 key cols");
  int64_t l = left_cols[0].numel();
  int64_t r = right_cols[0].numel();
  auto opts_long = left_cols[0].options().dtype(at::kLong);
  auto opts_int = left_cols[0].options().dtype(at::kInt);
  if (l == 0 || r == 0) {
    return {at::empty({0}, opts_long), at::empty({0}, opts_long)};
  }
  KeyCols probe{}, build{};
  probe.k = build.k = static_cast<int>(left_cols.size());
  for (size_t j = 0; j < left_cols.size(); ++j) {
    TORCH_CHECK(left_cols[j].is_cuda() && right_cols[j].is_cuda());
    TORCH_CHECK(left_cols[j].dtype() == at::kInt && right_cols[j].dtype() == at::kInt);
    probe.c[j] = left_cols[j].data_ptr<int32_t>();
    build.c[j] = right_cols[j].data_ptr<int32_t>();
  }
  auto stream = cur_stream();
  bool lds = r <= kHjLdsRows;
  at::Tensor heads, next;
  uint32_t mask = 0;
  if (lds) {
    // small build side: each block stages the chained table in LDS
    auto cnt = at::empty({l}, opts_int);
    hipLaunchKernelGGL(hj_count_lds, dim3(grid_for(l)), dim3(kBlock), 0,
                       stream, probe, l, build, r, cnt.data_ptr<int32_t>());
    HIP_OK(hipGetLastError());
    auto offs = at::cumsum(cnt, 0, at::kLong);
    int64_t total = offs[-1].item<int64_t>();
    auto offs_excl = offs - cnt.to(at::kLong);
    auto li = at::empty({total}, opts_long);
    auto ri = at::empty({total}, opts_long);
    if (total > 0) {
      hipLaunchKernelGGL(hj_emit_lds, dim3(grid_for(l)), dim3(kBlock), 0,
                         stream, probe, l, build, r,
                         offs_excl.data_ptr<int64_t>(),
                         li.data_ptr<int64_t>(), ri.data_ptr<int64_t>());
      HIP_OK(hipGetLastError());
    }
    return {li, ri};
  }
  // table size: next pow2 >= 2r
  uint64_t h = 1;
  while (h < static_cast<uint64_t>(2 * r)) h <<= 1;
  mask = static_cast<uint32_t>(h - 1);
  heads = at::full({static_cast<int64_t>(h)}, -1, opts_int);
  next = at::empty({r}, opts_int);
  hipLaunchKernelGGL(hj_build, dim3(grid_for(r)), dim3(kBlock), 0, stream,
                     build, r, mask, heads.data_ptr<int32_t>(),
                     next.data_ptr<int32_t>());
  HIP_OK(hipGetLastError());
  auto cnt = at::empty({l}, opts_int);
  hipLaunchKernelGGL(hj_count, dim3(grid_for(l)), dim3(kBlock), 0, stream,
                     probe, l, build, heads.data_ptr<int32_t>(),
                     next.data_ptr<int32_t>(), mask, cnt.data_ptr<int32_t>());
  HIP_OK(hipGetLastError());
  auto offs = at::cumsum(cnt, 0, at::kLong);
  int64_t total = offs[-1].item<int64_t>();
  auto offs_excl = offs - cnt.to(at::kLong);
  auto li = at::empty({total}, opts_long);
  auto ri = at::empty({total}, opts_long);
  if (total > 0) {
    hipLaunchKernelGGL(hj_emit, dim3(grid_for(l)), dim3(kBlock), 0, stream,
                       probe, l, build, heads.data_ptr<int32_t>(),
                       next.data_ptr<int32_t>(), mask,
                       offs_excl.data_ptr<int64_t>(), li.data_ptr<int64_t>(),
                       ri.data_ptr<int64_t>());
    HIP_OK(hipGetLastError());
  }
  return {li, ri};
}

// K5 filter bytecode evaluation -> bool mask.
at::Tensor filter_bytecode(at::Tensor ops, at::Tensor args, at::Tensor consts,
                           std::vector<at::Tensor> cols, at::Tensor value_col,
                           int64_t n_rows) {
  TORCH_CHECK(cols.size() <= static_cast<size_t>(kMaxCols),
              "filter: at most 16 columns");
  auto out = at::empty({n_rows},
                       value_col.options().dtype(at::kBool));
  if (n_rows == 0) return out;
  // bytecode is tiny: keep it device-resident (caller moves it)
  TORCH_CHECK(ops.is_cuda() && args.is_cuda() && consts.is_cuda());
  FilterProg prog{};
  prog.ops = ops.data_ptr<int32_t>();
  prog.args = args.data_ptr<int32_t>();
  prog.consts = consts.data_ptr<double>();
  prog.n_ops = static_cast<int>(ops.numel());
  BindCols bc{};
  bc.k = static_cast<int>(cols.size());
  for (size_t j = 0; j < cols.size(); ++j) {
    TORCH_CHECK(cols[j].is_cuda() && cols[j].dtype() == at::kInt);
    bc.c[j] = cols[j].data_ptr<int32_t>();
  }
  hipLaunchKernelGGL(filter_eval, dim3(grid_for(n_rows)), dim3(kBlock), 0,
                     cur_stream(), prog, bc, value_col.data_ptr<double>(),
                     value_col.numel(), n_rows, out.data_ptr<bool>());
  HIP_OK(hipGetLastError());
  return out;
}

// ------------------------------------------------- host bulk N-Triples parse
// Native replacement for the reference's crossbeam parse pipeline
// (sparql_database.rs:636-795): one pass tokenizes lines, interns terms in
// a local dictionary and emits local-id triples; Python merges the (much
// smaller) unique-string table into the main dictionary and remaps the id
// columns vectorized.  RDF-star lines (`<<`) fall back to the Python path.
static inline std::string unescape_nt(const char* s, size_t len) {
  std::string out;
  out.reserve(len);
  for (size_t i = 0; i < len; ++i) {
    char c = s[i];
    if (c == '\\' && i + 1 < len) {
      char n = s[i + 1];
      switch (n) {
        case 'n': out.push_back('\n'); ++i; continue;
        case 't': out.push_back('\t'); ++i; continue;
        case 'r': out.push_back('\r'); ++i; continue;
        case '"': out.push_back('"'); ++i; continue;
        case '\\': out.push_back('\\'); ++i; continue;
        case 'u':
          if (i + 5 < len) {
            unsigned cp = std::stoul(std::string(s + i + 2, 4), nullptr, 16);
            // minimal UTF-8 encode
            if (cp < 0x80) out.push_back(static_cast<char>(cp));
            else if (cp < 0x800) {
              out.push_back(static_cast<char>(0xC0 | (cp >> 6)));
              out.push_back(static_cast<char>(0x80 | (cp & 0x3F)));
            } else {
              out.push_back(static_cast<char>(0xE0 | (cp >> 12)));
              out.push_back(static_cast<char>(0x80 | ((cp >> 6) & 0x3F)));
              out.push_back(static_cast<char>(0x80 | (cp & 0x3F)));
            }
            i += 5;
            continue;
          }
          break;
        default: break;
      }
    }
    out.push_back(c);
  }
  return out;
}

py::tuple parse_nlines_host(const std::string& text, bool quads) {
  std::unordered_map<std::string, int64_t> interned;
  std::vector<std::string> strings;
  std::vector<int64_t> ids;           // n*3 local ids
  std::vector<int64_t> fallback;      // line numbers for the Python path
  ids.reserve(1 << 16);

  auto intern = [&](std::string&& s) -> int64_t {
    auto it = interned.find(s);
    if (it != interned.end()) return it->second;
    int64_t id = static_cast<int64_t>(strings.size());
    interned.emplace(s, id);
    strings.push_back(std::move(s));
    return id;
  };

  size_t pos = 0, line_no = 0;
  const size_t n = text.size();
  while (pos < n) {
    size_t eol = text.find('\n', pos);
    if (eol == std::string::npos) eol = n;
    const char* line = text.data() + pos;
    size_t len = eol - pos;
    size_t save_pos = pos;
    pos = eol + 1;
    ++line_no;
    // trim
    size_t b = 0, e = len;
    while (b < e && isspace(static_cast<unsigned char>(line[b]))) ++b;
    while (e > b && isspace(static_cast<unsigned char>(line[e - 1]))) --e;
    if (b >= e || line[b] == '#') continue;
    // RDF-star or exotic: Python fallback
    bool star = false;
    for (size_t i = b; i + 1 < e; ++i)
      if (line[i] == '<' && line[i + 1] == '<') { star = true; break; }
    if (star) { fallback.push_back(static_cast<int64_t>(line_no - 1)); continue; }
    int64_t term_ids[4];
    const int max_terms = quads ? 4 : 3;
    int nt = 0;
    size_t i = b;
    bool ok = true;
    while (i < e && nt < max_terms) {
      while (i < e && isspace(static_cast<unsigned char>(line[i]))) ++i;
      if (i >= e) break;
      char c = line[i];
      if (c == '<') {
        size_t j = i + 1;
        while (j < e && line[j] != '>') ++j;
        if (j >= e) { ok = false; break; }
        term_ids[nt++] = intern(unescape_nt(line + i + 1, j - i - 1));
        i = j + 1;
      } else if (c == '"') {
        size_t j = i + 1;
        while (j < e && !(line[j] == '"' && line[j - 1] != '\\')) ++j;
        if (j >= e) { ok = false; break; }
        term_ids[nt++] = intern(unescape_nt(line + i + 1, j - i - 1));
        i = j + 1;
        // skip @lang / ^^<datatype>
        while (i < e && !isspace(static_cast<unsigned char>(line[i]))
               && line[i] != '.') {
          if (line[i] == '<') { while (i < e && line[i] != '>') ++i; }
          ++i;
        }
      } else if (c == '_' ) {
        size_t j = i;
        while (j < e && !isspace(static_cast<unsigned char>(line[j]))) ++j;
        term_ids[nt++] = intern(std::string(line + i, j - i));
        i = j;
      } else if (c == '.') {
        break;
      } else {
        ok = false;
        break;
      }
    }
    if (!ok || nt < 3 || (!quads && nt != 3)) {
      fallback.push_back(static_cast<int64_t>(line_no - 1));
      continue;
    }
    ids.push_back(term_ids[0]);
    ids.push_back(term_ids[1]);
    ids.push_back(term_ids[2]);
    if (quads) ids.push_back(nt == 4 ? term_ids[3] : -1);
    (void)save_pos;
  }
  const int64_t width = quads ? 4 : 3;
  auto t = at::from_blob(ids.data(),
                         {static_cast<int64_t>(ids.size()) / width, width},
                         at::kLong).clone();
  py::list pystrings;
  for (auto& s : strings) pystrings.append(py::bytes(s));
  py::list pyfallback;
  for (auto f : fallback) pyfallback.append(f);
  return py::make_tuple(t, pystrings, pyfallback);
}

// ---- multithreaded bulk parse (replaces the reference's crossbeam
// worker pipeline, sparql_database.rs:630-804 / :1264-1463): the input
// splits at line boundaries into one chunk per thread, each thread runs
// the single-pass tokenizer with a chunk-local intern table, then the
// main thread merges the (small) unique-string tables and the chunk
// threads remap their id columns in parallel.  The GIL is released for
// the whole native phase.
#include <deque>
#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

constexpr int kVocabShards = 32;

// open-addressing string_view -> id map (linear probing, stored 64-bit
// hash, power-of-two).  A node-based unordered_map pays one malloc plus
// a pointer chase per entry — at the 100M-triple bulk-load's ~43M unique
// terms (and ~130M chunk-local interns) that allocator traffic dominated
// the parse and merge phases.
struct FlatMap {
  struct Slot {
    const char* ptr = nullptr;   // nullptr => empty
    uint64_t hash = 0;
    int64_t id = 0;
    uint32_t len = 0;
  };
  std::vector<Slot> slots;
  size_t mask;
  size_t count = 0;
  FlatMap() : slots(1024), mask(1023) {}
  void grow() {
    std::vector<Slot> old = std::move(slots);
    size_t nsz = (mask + 1) * 2;
    slots.assign(nsz, Slot{});
    mask = nsz - 1;
    for (auto& s0 : old)
      if (s0.ptr) {
        size_t i = s0.hash & mask;
        while (slots[i].ptr) i = (i + 1) & mask;
        slots[i] = s0;
      }
  }
  // returns the slot index for key (h, sv); found=false means the slot
  // was claimed fresh — caller must fill ptr (e.g. repoint to an arena
  // copy) and id.  Grows BEFORE probing, so a returned index is valid
  // until the next find_or_insert.
  size_t find_or_insert(uint64_t h, std::string_view sv, bool& found) {
    if ((count + 1) * 10 >= (mask + 1) * 7) grow();
    size_t i = h & mask;
    for (;;) {
      Slot& sl = slots[i];
      if (sl.ptr == nullptr) {
        sl.ptr = sv.data();
        sl.hash = h;
        sl.len = static_cast<uint32_t>(sv.size());
        ++count;
        found = false;
        return i;
      }
      if (sl.hash == h && sl.len == sv.size()
          && memcmp(sl.ptr, sv.data(), sl.len) == 0) {
        found = true;
        return i;
      }
      i = (i + 1) & mask;
    }
  }
  // SIZE_MAX if absent
  size_t find(uint64_t h, std::string_view sv) const {
    size_t i = h & mask;
    for (;;) {
      const Slot& sl = slots[i];
      if (sl.ptr == nullptr) return SIZE_MAX;
      if (sl.hash == h && sl.len == sv.size()
          && memcmp(sl.ptr, sv.data(), sl.len) == 0) return i;
      i = (i + 1) & mask;
    }
  }
};

struct ParseChunkOut {
  // term views point into the shared input text (zero-copy for the
  // overwhelmingly common escape-free terms) or into `owned` (deque:
  // stable addresses) for unescaped materializations
  FlatMap interned;
  std::vector<std::string_view> views;
  std::vector<uint64_t> hashes;    // std::hash of each view, precomputed
  // view indices bucketed by hash % kVocabShards at parse time, so each
  // vocab-merge shard thread walks only its own items instead of
  // filter-scanning every chunk's full hash array
  std::vector<uint32_t> by_shard[kVocabShards];
  std::deque<std::string> owned;
  std::vector<int64_t> ids;
  std::vector<int64_t> fallback;   // chunk-local line numbers
  int64_t line_count = 0;
};

static void parse_chunk_nt(const char* data, size_t begin, size_t end,
                           bool quads, ParseChunkOut& out) {
  std::hash<std::string_view> sv_hasher;
  auto intern_sv = [&](std::string_view sv) -> int64_t {
    uint64_t hh = sv_hasher(sv);
    bool found;
    size_t si = out.interned.find_or_insert(hh, sv, found);
    if (found) return out.interned.slots[si].id;
    int64_t id = static_cast<int64_t>(out.views.size());
    out.interned.slots[si].id = id;
    out.views.push_back(sv);
    out.hashes.push_back(hh);
    out.by_shard[hh % kVocabShards].push_back(static_cast<uint32_t>(id));
    return id;
  };
  auto intern = [&](const char* ptr, size_t len) -> int64_t {
    if (memchr(ptr, '\\', len) == nullptr)
      return intern_sv(std::string_view(ptr, len));
    std::string u = unescape_nt(ptr, len);
    uint64_t hh = sv_hasher(std::string_view(u));
    bool found;
    size_t si = out.interned.find_or_insert(hh, std::string_view(u), found);
    if (found) return out.interned.slots[si].id;
    out.owned.push_back(std::move(u));
    std::string_view sv(out.owned.back());
    // the probe keyed on the TEMPORARY string — repoint to the stable copy
    out.interned.slots[si].ptr = sv.data();
    int64_t id = static_cast<int64_t>(out.views.size());
    out.interned.slots[si].id = id;
    out.views.push_back(sv);
    out.hashes.push_back(hh);
    out.by_shard[hh % kVocabShards].push_back(static_cast<uint32_t>(id));
    return id;
  };
  size_t pos = begin;
  int64_t line_no = 0;
  while (pos < end) {
    const char* nl = static_cast<const char*>(
        memchr(data + pos, '\n', end - pos));
    size_t eol = nl ? static_cast<size_t>(nl - data) : end;
    const char* line = data + pos;
    size_t len = eol - pos;
    pos = eol + 1;
    ++line_no;
    size_t b = 0, e = len;
    while (b < e && isspace(static_cast<unsigned char>(line[b]))) ++b;
    while (e > b && isspace(static_cast<unsigned char>(line[e - 1]))) --e;
    if (b >= e || line[b] == '#') continue;
    // RDF-star detection via memchr hops (the naive per-byte scan read
    // every line twice; memchr skips to the handful of '<' per line)
    bool star = false;
    for (const char* q = line + b;
         (q = static_cast<const char*>(
              memchr(q, '<', static_cast<size_t>(line + e - q)))) != nullptr;
         ++q) {
      if (q + 1 < line + e && q[1] == '<') { star = true; break; }
    }
    if (star) { out.fallback.push_back(line_no - 1); continue; }
    int64_t term_ids[4];
    const int max_terms = quads ? 4 : 3;
    int nt = 0;
    size_t i = b;
    bool ok = true;
    while (i < e && nt < max_terms) {
      while (i < e && isspace(static_cast<unsigned char>(line[i]))) ++i;
      if (i >= e) break;
      char c = line[i];
      if (c == '<') {
        const char* gt = static_cast<const char*>(
            memchr(line + i + 1, '>', e - i - 1));
        if (gt == nullptr) { ok = false; break; }
        size_t j = static_cast<size_t>(gt - line);
        term_ids[nt++] = intern(line + i + 1, j - i - 1);
        i = j + 1;
      } else if (c == '"') {
        size_t j = i + 1;
        for (;;) {
          const char* qq = static_cast<const char*>(
              memchr(line + j, '"', e - j));
          if (qq == nullptr) { j = e; break; }
          j = static_cast<size_t>(qq - line);
          if (line[j - 1] != '\\') break;
          ++j;
        }
        if (j >= e) { ok = false; break; }
        term_ids[nt++] = intern(line + i + 1, j - i - 1);
        i = j + 1;
        while (i < e && !isspace(static_cast<unsigned char>(line[i]))
               && line[i] != '.') {
          if (line[i] == '<') { while (i < e && line[i] != '>') ++i; }
          ++i;
        }
      } else if (c == '_') {
        size_t j = i;
        while (j < e && !isspace(static_cast<unsigned char>(line[j]))) ++j;
        term_ids[nt++] = intern_sv(std::string_view(line + i, j - i));
        i = j;
      } else if (c == '.') {
        break;
      } else {
        ok = false;
        break;
      }
    }
    if (!ok || nt < 3 || (!quads && nt != 3)) {
      out.fallback.push_back(line_no - 1);
      continue;
    }
    out.ids.push_back(term_ids[0]);
    out.ids.push_back(term_ids[1]);
    out.ids.push_back(term_ids[2]);
    if (quads) out.ids.push_back(nt == 4 ? term_ids[3] : -1);
  }
  out.line_count = line_no;
}

py::tuple parse_nlines_host_mt(const std::string& text, bool quads,
                               int64_t n_threads) {
  const size_t n = text.size();
  int nt = static_cast<int>(n_threads);
  if (nt <= 0) {
    nt = static_cast<int>(std::thread::hardware_concurrency());
    if (nt <= 0) nt = 8;
  }
  if (nt > 32) nt = 32;
  if (n < (1 << 20)) nt = 1;
  // chunk boundaries at line starts
  std::vector<size_t> starts(1, 0);
  for (int i = 1; i < nt; ++i) {
    size_t p = n * static_cast<size_t>(i) / nt;
    const char* nl = static_cast<const char*>(
        memchr(text.data() + p, '\n', n - p));
    starts.push_back(nl ? static_cast<size_t>(nl - text.data()) + 1 : n);
  }
  starts.push_back(n);
  std::vector<ParseChunkOut> outs(nt);
  std::vector<std::string> g_strings;
  std::vector<std::vector<int64_t>> remaps(nt);
  std::vector<int64_t> line_off(nt + 1, 0);
  int64_t total_rows = 0, total_fb = 0;
  {
    py::gil_scoped_release release;
    std::vector<std::thread> threads;
    for (int i = 0; i < nt; ++i)
      threads.emplace_back(parse_chunk_nt, text.data(), starts[i],
                           starts[i + 1], quads, std::ref(outs[i]));
    for (auto& t : threads) t.join();
    // merge unique-string tables (small relative to the line count)
    size_t uniq_upper0 = 0;
    for (int i = 0; i < nt; ++i) uniq_upper0 += outs[i].views.size();
    g_strings.reserve(uniq_upper0);
    std::unordered_map<std::string_view, int64_t> global;
    global.reserve(uniq_upper0 * 2);
    for (int i = 0; i < nt; ++i) {
      remaps[i].resize(outs[i].views.size());
      for (size_t k = 0; k < outs[i].views.size(); ++k) {
        auto it = global.find(outs[i].views[k]);
        if (it == global.end()) {
          int64_t id = static_cast<int64_t>(g_strings.size());
          g_strings.emplace_back(outs[i].views[k]);
          global.emplace(std::string_view(g_strings.back()), id);
          remaps[i][k] = id;
        } else {
          remaps[i][k] = it->second;
        }
      }
      line_off[i + 1] = line_off[i] + outs[i].line_count;
      total_rows += static_cast<int64_t>(outs[i].ids.size());
      total_fb += static_cast<int64_t>(outs[i].fallback.size());
    }
    // remap chunk-local ids to global ids, in parallel
    threads.clear();
    for (int i = 0; i < nt; ++i)
      threads.emplace_back([&, i]() {
        auto& rm = remaps[i];
        for (auto& v : outs[i].ids) v = rm[v];
      });
    for (auto& t : threads) t.join();
  }
  const int64_t width = quads ? 4 : 3;
  auto ids_t = at::empty({total_rows / width, width}, at::kLong);
  int64_t off = 0;
  int64_t* dst = ids_t.data_ptr<int64_t>();
  for (int i = 0; i < nt; ++i) {
    if (!outs[i].ids.empty()) {
      memcpy(dst + off, outs[i].ids.data(),
             outs[i].ids.size() * sizeof(int64_t));
      off += static_cast<int64_t>(outs[i].ids.size());
    }
  }
  py::list pystrings;
  for (auto& s : g_strings) pystrings.append(py::bytes(s));
  py::list pyfallback;
  for (int i = 0; i < nt; ++i)
    for (auto f : outs[i].fallback) pyfallback.append(f + line_off[i]);
  return py::make_tuple(ids_t, pystrings, pyfallback);
}

py::tuple parse_ntriples_host_mt(const std::string& text,
                                 int64_t n_threads) {
  return parse_nlines_host_mt(text, false, n_threads);
}

py::tuple parse_nquads_host_mt(const std::string& text, int64_t n_threads) {
  return parse_nlines_host_mt(text, true, n_threads);
}

// end-to-end file ingest: read + chunk-per-thread parse (GIL released),
// merge intern tables, then ONE GIL pass interns the UNIQUE strings
// straight into the Python Dictionary containers (str_to_id dict /
// id_to_str list / values list — no intermediate py::bytes round trip),
// and the id columns remap to global dictionary ids in parallel.
// Returns (int32 [n,3] global-id rows, fallback line numbers).
py::tuple parse_ntriples_file_encode(const std::string& path,
                                     int64_t n_threads,
                                     py::dict str_to_id, py::list id_to_str,
                                     py::list values, int64_t max_id) {
  const bool dbg = getenv("KOLIBRIE_PARSE_DEBUG") != nullptr;
  auto tick = std::chrono::steady_clock::now();
  auto lap = [&](const char* what) {
    if (!dbg) return;
    auto now = std::chrono::steady_clock::now();
    fprintf(stderr, "[parse] %s: %.3fs\n", what,
            std::chrono::duration<double>(now - tick).count());
    tick = now;
  };
  std::string text;
  {
    py::gil_scoped_release release;
    FILE* f = fopen(path.c_str(), "rb");
    TORCH_CHECK(f != nullptr, "cannot open ", path);
    fseek(f, 0, SEEK_END);
    long sz = ftell(f);
    fseek(f, 0, SEEK_SET);
    text.resize(static_cast<size_t>(sz));
    size_t rd = fread(text.data(), 1, static_cast<size_t>(sz), f);
    fclose(f);
    TORCH_CHECK(rd == static_cast<size_t>(sz), "short read of ", path);
  }
  lap("read");
  const size_t n = text.size();
  int nt = static_cast<int>(n_threads);
  if (nt <= 0) {
    nt = static_cast<int>(std::thread::hardware_concurrency());
    if (nt <= 0) nt = 8;
  }
  if (nt > 32) nt = 32;
  if (n < (1 << 20)) nt = 1;
  std::vector<size_t> starts(1, 0);
  for (int i = 1; i < nt; ++i) {
    size_t p = n * static_cast<size_t>(i) / nt;
    const char* nl = static_cast<const char*>(
        memchr(text.data() + p, '\n', n - p));
    starts.push_back(nl ? static_cast<size_t>(nl - text.data()) + 1 : n);
  }
  starts.push_back(n);
  std::vector<ParseChunkOut> outs(nt);
  std::vector<std::vector<std::string_view>> shard_views;
  std::vector<std::vector<int64_t>> remaps(nt);
  std::vector<int64_t> line_off(nt + 1, 0);
  int64_t total_rows = 0;
  {
    py::gil_scoped_release release;
    std::vector<std::thread> threads;
    for (int i = 0; i < nt; ++i)
      threads.emplace_back(parse_chunk_nt, text.data(), starts[i],
                           starts[i + 1], false, std::ref(outs[i]));
    for (auto& t : threads) t.join();
    if (dbg) { py::gil_scoped_acquire a; lap("parallel parse"); }
    // SHARDED parallel merge: shard strings by hash; each thread owns one
    // shard, scans every chunk's view table and dedups only its shard
    // (remap entries encode (shard, local id); resolved after the GIL
    // interning pass assigns Python ids per shard)
    size_t uniq_upper = 0;
    for (int i = 0; i < nt; ++i) {
      remaps[i].resize(outs[i].views.size());
      uniq_upper += outs[i].views.size();
      line_off[i + 1] = line_off[i] + outs[i].line_count;
      total_rows += static_cast<int64_t>(outs[i].ids.size());
    }
    shard_views.resize(nt);
    {
      std::vector<std::thread> mthreads;
      for (int t = 0; t < nt; ++t)
        mthreads.emplace_back([&, t]() {
          std::unordered_map<std::string_view, int64_t> local;
          local.reserve(uniq_upper / std::max(1, nt) * 2 + 64);
          auto& sv = shard_views[t];
          for (int i = 0; i < nt; ++i) {
            auto& views = outs[i].views;
            auto& hs = outs[i].hashes;
            auto& rm = remaps[i];
            for (size_t k = 0; k < views.size(); ++k) {
              if (static_cast<int>(hs[k] % nt) != t) continue;
              auto it = local.find(views[k]);
              int64_t L;
              if (it == local.end()) {
                L = static_cast<int64_t>(sv.size());
                local.emplace(views[k], L);
                sv.push_back(views[k]);
              } else {
                L = it->second;
              }
              rm[k] = (static_cast<int64_t>(t) << 40) | L;
            }
          }
        });
      for (auto& t : mthreads) t.join();
    }
    if (dbg) { py::gil_scoped_acquire a; lap("merge"); }
  }
  // GIL pass: intern each shard's unique strings into the Python
  // dictionary (one SetDefault per string; a shared 0.0 float object for
  // the non-numeric majority)
  std::vector<std::vector<int64_t>> py_ids(shard_views.size());
  {
    PyObject* d = str_to_id.ptr();
    PyObject* lst = id_to_str.ptr();
    PyObject* vals = values.ptr();
    PyObject* zero = PyFloat_FromDouble(0.0);
    int64_t next_id = static_cast<int64_t>(PyList_GET_SIZE(lst));
    char numbuf[64];
    for (size_t t = 0; t < shard_views.size(); ++t) {
      auto& sv = shard_views[t];
      auto& pid = py_ids[t];
      pid.resize(sv.size());
      for (size_t k = 0; k < sv.size(); ++k) {
        std::string_view s = sv[k];
        PyObject* key = PyUnicode_DecodeUTF8(
            s.data(), static_cast<Py_ssize_t>(s.size()), "replace");
        TORCH_CHECK(key != nullptr, "utf-8 decode failed");
        TORCH_CHECK(next_id < max_id, "dictionary ID space exhausted");
        PyObject* idobj = PyLong_FromLongLong(next_id);
        PyObject* prev = PyDict_SetDefault(d, key, idobj);  // borrowed
        if (prev != idobj) {
          pid[k] = PyLong_AsLongLong(prev);
          Py_DECREF(idobj);
          Py_DECREF(key);
          continue;
        }
        PyList_Append(lst, key);
        // numeric value column: full-consume strtod, non-finite -> 0.0
        double v = 0.0;
        char c0 = s.empty() ? 0 : s[0];
        if (((c0 >= '0' && c0 <= '9') || c0 == '-' || c0 == '+'
             || c0 == '.' || c0 == ' ' || c0 == '\t'
             || c0 == 'i' || c0 == 'I' || c0 == 'n' || c0 == 'N')
            && s.size() < sizeof(numbuf)) {
          memcpy(numbuf, s.data(), s.size());
          numbuf[s.size()] = '\0';
          char* endp = nullptr;
          double parsed = strtod(numbuf, &endp);
          if (endp != numbuf && endp != nullptr) {
            while (*endp == ' ' || *endp == '\t') ++endp;
            if (*endp == '\0' && std::isfinite(parsed)) v = parsed;
          }
        }
        if (v == 0.0) {
          PyList_Append(vals, zero);
        } else {
          PyObject* vobj = PyFloat_FromDouble(v);
          PyList_Append(vals, vobj);
          Py_DECREF(vobj);
        }
        Py_DECREF(idobj);
        Py_DECREF(key);
        pid[k] = next_id++;
      }
    }
    Py_DECREF(zero);
  }
  lap("dict intern");
  auto ids_t = at::empty({total_rows / 3, 3}, at::kInt);
  {
    py::gil_scoped_release release;
    std::vector<std::thread> threads;
    std::vector<int64_t> offs(nt + 1, 0);
    for (int i = 0; i < nt; ++i)
      offs[i + 1] = offs[i] + static_cast<int64_t>(outs[i].ids.size());
    int32_t* dst = ids_t.data_ptr<int32_t>();
    for (int i = 0; i < nt; ++i)
      threads.emplace_back([&, i]() {
        auto& rm = remaps[i];
        int32_t* d = dst + offs[i];
        for (size_t k = 0; k < outs[i].ids.size(); ++k) {
          int64_t e = rm[outs[i].ids[k]];
          d[k] = static_cast<int32_t>(static_cast<uint32_t>(
              py_ids[e >> 40][e & ((1LL << 40) - 1)]));
        }
      });
    for (auto& t : threads) t.join();
  }
  lap("remap");
  py::list pyfallback;
  for (int i = 0; i < nt; ++i)
    for (auto f : outs[i].fallback) pyfallback.append(f + line_off[i]);
  return py::make_tuple(ids_t, pyfallback);
}

// ------------------------------------------------------------ vocab annex
// Native bulk-vocabulary store (the C++-primary dictionary tail): once a
// database bulk-loads, the Python dictionary's prefix [0, P0) freezes and
// EVERY later term id lives here — sharded string->id maps with stable
// deque arenas, a by-id pointer index, and the float64 value column.
// Python's Dictionary consults it on lookup/encode/decode misses; result
// decode uses the batch exporter so only requested ids materialize as
// Python strings.
// append-only bump arena: one memcpy per interned string instead of a
// per-string malloc (a deque<std::string> arena paid ~43M heap
// allocations on a 100M-triple load — the dominant merge cost)
struct BumpArena {
  std::vector<std::unique_ptr<char[]>> blocks;
  size_t used = 0, cap = 0;
  std::string_view add(std::string_view sv) {
    if (blocks.empty() || used + sv.size() > cap) {
      // blocks.empty() guard: a zero-length first insert (e.g. the empty
      // literal "") must still allocate before blocks.back()
      size_t bs = std::max<size_t>(size_t(1) << 24, sv.size());
      blocks.emplace_back(new char[bs]);
      used = 0;
      cap = bs;
    }
    char* dst = blocks.back().get() + used;
    memcpy(dst, sv.data(), sv.size());
    used += sv.size();
    return std::string_view(dst, sv.size());
  }
};

struct VocabShard {
  FlatMap map;
  BumpArena arena;
};

struct Vocab {
  int nshards = kVocabShards;
  std::vector<VocabShard> shards;
  std::vector<std::string_view> by_id;  // index = id - base
  std::vector<double> values;
  int64_t base = -1;
  Vocab() : shards(nshards) {}
};
static std::vector<std::unique_ptr<Vocab>> g_vocabs;

static double parse_value_full(std::string_view s) {
  char c0 = s.empty() ? 0 : s[0];
  if (!((c0 >= '0' && c0 <= '9') || c0 == '-' || c0 == '+' || c0 == '.'
        || c0 == ' ' || c0 == '\t' || c0 == 'i' || c0 == 'I'
        || c0 == 'n' || c0 == 'N'))
    return 0.0;
  if (s.size() >= 63) return 0.0;
  char buf[64];
  memcpy(buf, s.data(), s.size());
  buf[s.size()] = '\0';
  char* endp = nullptr;
  double parsed = strtod(buf, &endp);
  if (endp == buf || endp == nullptr) return 0.0;
  while (*endp == ' ' || *endp == '\t') ++endp;
  if (*endp != '\0' || !std::isfinite(parsed)) return 0.0;
  return parsed;
}

int64_t vocab_create(int64_t base_id) {
  auto v = std::make_unique<Vocab>();
  v->base = base_id;
  g_vocabs.push_back(std::move(v));
  return static_cast<int64_t>(g_vocabs.size() - 1);
}

// create seeded with the Python dictionary's frozen prefix: the bulk
// parser dedups against these ids too (without this a prefix string —
// including "" at id 0, or any term interned before the first bulk
// load — would get a DUPLICATE annex id and constant-term queries
// would miss the bulk-loaded rows)
int64_t vocab_create_seeded(py::list strings) {
  auto v = std::make_unique<Vocab>();
  v->base = static_cast<int64_t>(strings.size());
  std::hash<std::string_view> hasher;
  int64_t i = 0;
  for (auto h : strings) {
    std::string str = py::cast<std::string>(h);
    std::string_view key(str);
    uint64_t hh = hasher(key);
    auto& sh = v->shards[hh % v->nshards];
    bool found;
    size_t si = sh.map.find_or_insert(hh, key, found);
    if (!found) {
      std::string_view sv = sh.arena.add(key);
      sh.map.slots[si].ptr = sv.data();
      sh.map.slots[si].id = i;
    }
    ++i;
  }
  g_vocabs.push_back(std::move(v));
  return static_cast<int64_t>(g_vocabs.size() - 1);
}

int64_t vocab_len(int64_t h) {
  return static_cast<int64_t>(g_vocabs.at(h)->by_id.size());
}

int64_t vocab_lookup(int64_t h, const std::string& s) {
  Vocab& v = *g_vocabs.at(h);
  std::hash<std::string_view> hasher;
  uint64_t hh = hasher(std::string_view(s));
  auto& sh = v.shards[hh % v.nshards];
  size_t si = sh.map.find(hh, std::string_view(s));
  return si == SIZE_MAX ? -1 : sh.map.slots[si].id;
}

int64_t vocab_insert(int64_t h, const std::string& s) {
  Vocab& v = *g_vocabs.at(h);
  std::hash<std::string_view> hasher;
  uint64_t hh = hasher(std::string_view(s));
  auto& sh = v.shards[hh % v.nshards];
  bool found;
  size_t si = sh.map.find_or_insert(hh, std::string_view(s), found);
  if (found) return sh.map.slots[si].id;
  std::string_view sv = sh.arena.add(std::string_view(s));
  sh.map.slots[si].ptr = sv.data();  // repoint off the caller's buffer
  int64_t id = v.base + static_cast<int64_t>(v.by_id.size());
  sh.map.slots[si].id = id;
  v.by_id.push_back(sv);
  v.values.push_back(parse_value_full(sv));
  return id;
}

py::object vocab_get(int64_t h, int64_t id) {
  Vocab& v = *g_vocabs.at(h);
  int64_t k = id - v.base;
  if (k < 0 || k >= static_cast<int64_t>(v.by_id.size()))
    return py::none();
  std::string_view s = v.by_id[k];
  PyObject* u = PyUnicode_DecodeUTF8(s.data(),
                                     static_cast<Py_ssize_t>(s.size()),
                                     "replace");
  return py::reinterpret_steal<py::object>(u);
}

// batch decode for result materialization: CPU int64 id tensor -> Python
// list of str (ids outside the annex range decode to None)
py::list vocab_decode_batch(int64_t h, at::Tensor ids) {
  Vocab& v = *g_vocabs.at(h);
  TORCH_CHECK(!ids.is_cuda() && ids.dtype() == at::kLong);
  const int64_t* r = ids.data_ptr<int64_t>();
  int64_t n = ids.numel();
  py::list out(n);
  for (int64_t i = 0; i < n; ++i) {
    int64_t k = r[i] - v.base;
    PyObject* o;
    if (k < 0 || k >= static_cast<int64_t>(v.by_id.size())) {
      o = Py_None;
      Py_INCREF(o);
    } else {
      std::string_view s = v.by_id[k];
      o = PyUnicode_DecodeUTF8(s.data(),
                               static_cast<Py_ssize_t>(s.size()), "replace");
      if (o == nullptr) { o = Py_None; Py_INCREF(o); }
    }
    PyList_SET_ITEM(out.ptr(), i, o);
  }
  return out;
}

double vocab_value(int64_t h, int64_t id) {
  Vocab& v = *g_vocabs.at(h);
  int64_t k = id - v.base;
  if (k < 0 || k >= static_cast<int64_t>(v.values.size())) return 0.0;
  return v.values[k];
}

at::Tensor vocab_values(int64_t h) {
  Vocab& v = *g_vocabs.at(h);
  return at::from_blob(v.values.data(),
                       {static_cast<int64_t>(v.values.size())},
                       at::kDouble).clone();
}

py::list vocab_export_strings(int64_t h, int64_t start, int64_t count) {
  Vocab& v = *g_vocabs.at(h);
  py::list out;
  int64_t n = static_cast<int64_t>(v.by_id.size());
  for (int64_t k = start; k < std::min(start + count, n); ++k) {
    std::string_view s = v.by_id[k];
    PyObject* u = PyUnicode_DecodeUTF8(
        s.data(), static_cast<Py_ssize_t>(s.size()), "replace");
    out.append(py::reinterpret_steal<py::object>(u));
  }
  return out;
}

// file -> parse -> intern DIRECTLY into the vocab annex (no Python-object
// pass at all); returns (int32 [n,3] global-id rows, fallback lines)
py::tuple parse_ntriples_file_annex(const std::string& path,
                                    int64_t n_threads, int64_t h,
                                    int64_t max_id) {
  const bool dbg = getenv("KOLIBRIE_PARSE_DEBUG") != nullptr;
  auto tick = std::chrono::steady_clock::now();
  auto lap = [&](const char* what) {
    if (!dbg) return;
    auto now = std::chrono::steady_clock::now();
    fprintf(stderr, "[annex] %s: %.3fs\n", what,
            std::chrono::duration<double>(now - tick).count());
    tick = now;
  };
  Vocab& voc = *g_vocabs.at(h);
  // the bulky scratch (10 GB file text + per-chunk intern maps with tens
  // of millions of nodes) lives on the heap and is freed by a DETACHED
  // reaper thread after return — its node-by-node teardown takes seconds
  // and would otherwise serialize at scope exit, after all useful work
  struct AnnexScratch {
    const char* map_ptr = nullptr;   // mmap'd input (preferred: no copy)
    size_t map_len = 0;
    std::string text;                // fallback buffer if mmap fails
    std::vector<ParseChunkOut> outs;
    std::vector<std::vector<int64_t>> remaps;
    ~AnnexScratch() {
      if (map_ptr)
        munmap(const_cast<char*>(map_ptr), map_len);
    }
  };
  auto scratch = std::make_unique<AnnexScratch>();
  {
    py::gil_scoped_release release;
    int fd = open(path.c_str(), O_RDONLY);
    TORCH_CHECK(fd >= 0, "cannot open ", path);
    struct stat st {};
    TORCH_CHECK(fstat(fd, &st) == 0, "cannot stat ", path);
    size_t sz = static_cast<size_t>(st.st_size);
    void* m = sz ? mmap(nullptr, sz, PROT_READ, MAP_PRIVATE, fd, 0)
                 : nullptr;
    if (m != nullptr && m != MAP_FAILED) {
      madvise(m, sz, MADV_WILLNEED);
      scratch->map_ptr = static_cast<const char*>(m);
      scratch->map_len = sz;
    } else if (sz) {
      scratch->text.resize(sz);
      size_t off = 0;
      while (off < sz) {
        ssize_t rd = read(fd, scratch->text.data() + off, sz - off);
        TORCH_CHECK(rd > 0, "short read of ", path);
        off += static_cast<size_t>(rd);
      }
    }
    close(fd);
  }
  const char* data = scratch->map_ptr ? scratch->map_ptr
                                      : scratch->text.data();
  lap("read");
  const size_t n = scratch->map_ptr ? scratch->map_len
                                    : scratch->text.size();
  int nt = static_cast<int>(n_threads);
  if (nt <= 0) {
    nt = static_cast<int>(std::thread::hardware_concurrency());
    if (nt <= 0) nt = 8;
  }
  if (nt > 32) nt = 32;
  if (n < (1 << 20)) nt = 1;
  std::vector<size_t> starts(1, 0);
  for (int i = 1; i < nt; ++i) {
    size_t p = n * static_cast<size_t>(i) / nt;
    const char* nl = static_cast<const char*>(
        memchr(data + p, '\n', n - p));
    starts.push_back(nl ? static_cast<size_t>(nl - data) + 1 : n);
  }
  starts.push_back(n);
  scratch->outs.resize(nt);
  scratch->remaps.resize(nt);
  std::vector<ParseChunkOut>& outs = scratch->outs;
  std::vector<std::vector<int64_t>>& remaps = scratch->remaps;
  std::vector<int64_t> line_off(nt + 1, 0);
  int64_t total_rows = 0;
  at::Tensor ids_t;
  {
    py::gil_scoped_release release;
    std::vector<std::thread> threads;
    for (int i = 0; i < nt; ++i)
      threads.emplace_back(parse_chunk_nt, data, starts[i],
                           starts[i + 1], false, std::ref(outs[i]));
    for (auto& t : threads) t.join();
    if (dbg) { py::gil_scoped_acquire a; lap("parallel parse"); }
    for (int i = 0; i < nt; ++i) {
      remaps[i].resize(outs[i].views.size());
      line_off[i + 1] = line_off[i] + outs[i].line_count;
      total_rows += static_cast<int64_t>(outs[i].ids.size());
    }
    // sharded merge straight into the vocab: the vocab's shard count is
    // the parallelism; thread t owns vocab shard t.  Two phases: dedup
    // with PROVISIONAL (shard, local) ids, then per-shard contiguous
    // real-id blocks from prefix sums — no contended atomic, and ids are
    // deterministic for a given input + shard count.
    const int S = voc.nshards;
    std::vector<std::thread> mthreads;
    std::vector<std::vector<std::string_view>> pending(S);
    const int64_t kProv = 1LL << 62;  // provisional marker
    for (int t = 0; t < S; ++t)
      mthreads.emplace_back([&, t]() {
        auto& sh = voc.shards[t];
        auto& pend = pending[t];
        for (int i = 0; i < nt; ++i) {
          auto& views = outs[i].views;
          auto& rm = remaps[i];
          auto& hs = outs[i].hashes;
          for (uint32_t k : outs[i].by_shard[t]) {
            bool found;
            size_t si = sh.map.find_or_insert(hs[k], views[k], found);
            if (found) {
              rm[k] = sh.map.slots[si].id;  // may itself be provisional
              continue;
            }
            std::string_view sv = sh.arena.add(views[k]);
            sh.map.slots[si].ptr = sv.data();  // text/owned die with scratch
            int64_t prov = kProv | (static_cast<int64_t>(t) << 40)
                           | static_cast<int64_t>(pend.size());
            sh.map.slots[si].id = prov;
            pend.push_back(sv);
            rm[k] = prov;
          }
        }
      });
    for (auto& t : mthreads) t.join();
    if (dbg) { py::gil_scoped_acquire a; lap("vocab merge"); }
    // prefix sums -> real id blocks; fix the maps + fill by_id/values
    int64_t old_n = static_cast<int64_t>(voc.by_id.size());
    std::vector<int64_t> shard_base(S + 1, 0);
    for (int t = 0; t < S; ++t)
      shard_base[t + 1] = shard_base[t]
                          + static_cast<int64_t>(pending[t].size());
    int64_t new_total = old_n + shard_base[S];
    TORCH_CHECK(voc.base + new_total <= max_id,
                "dictionary ID space exhausted");
    voc.by_id.resize(new_total);
    voc.values.resize(new_total, 0.0);
    std::vector<std::thread> vthreads;
    for (int t = 0; t < S; ++t)
      vthreads.emplace_back([&, t]() {
        auto& sh = voc.shards[t];
        int64_t base = voc.base + old_n + shard_base[t];
        std::hash<std::string_view> hasher;
        for (size_t j = 0; j < pending[t].size(); ++j) {
          std::string_view sv = pending[t][j];
          int64_t id = base + static_cast<int64_t>(j);
          size_t si = sh.map.find(hasher(sv), sv);
          sh.map.slots[si].id = id;
          int64_t k = id - voc.base;
          voc.by_id[k] = sv;
          voc.values[k] = parse_value_full(sv);
        }
      });
    for (auto& t : vthreads) t.join();
    // resolve provisional remap entries to real ids
    std::vector<std::thread> fthreads;
    for (int i = 0; i < nt; ++i)
      fthreads.emplace_back([&, i]() {
        for (auto& e : remaps[i]) {
          if (e & kProv) {
            int64_t t = (e >> 40) & 0x3FFFFF;
            int64_t j = e & ((1LL << 40) - 1);
            e = voc.base + old_n + shard_base[t] + j;
          }
        }
      });
    for (auto& t : fthreads) t.join();
    // remap rows to global ids, in parallel
    ids_t = at::empty({total_rows / 3, 3}, at::kInt);
    std::vector<int64_t> offs(nt + 1, 0);
    for (int i = 0; i < nt; ++i)
      offs[i + 1] = offs[i] + static_cast<int64_t>(outs[i].ids.size());
    int32_t* dst = ids_t.data_ptr<int32_t>();
    std::vector<std::thread> rthreads;
    for (int i = 0; i < nt; ++i)
      rthreads.emplace_back([&, i]() {
        auto& rm = remaps[i];
        int32_t* d = dst + offs[i];
        for (size_t k = 0; k < outs[i].ids.size(); ++k)
          d[k] = static_cast<int32_t>(
              static_cast<uint32_t>(rm[outs[i].ids[k]]));
      });
    for (auto& t : rthreads) t.join();
    if (dbg) { py::gil_scoped_acquire a; lap("register+remap"); }
  }
  py::list pyfallback;
  for (int i = 0; i < nt; ++i)
    for (auto f : outs[i].fallback) pyfallback.append(f + line_off[i]);
  std::thread([sc = scratch.release()]() { delete sc; }).detach();
  return py::make_tuple(ids_t, pyfallback);
}

py::tuple parse_ntriples_file_mt(const std::string& path,
                                 int64_t n_threads) {
  std::string text;
  {
    py::gil_scoped_release release;
    FILE* f = fopen(path.c_str(), "rb");
    TORCH_CHECK(f != nullptr, "cannot open ", path);
    fseek(f, 0, SEEK_END);
    long sz = ftell(f);
    fseek(f, 0, SEEK_SET);
    text.resize(static_cast<size_t>(sz));
    size_t rd = fread(text.data(), 1, static_cast<size_t>(sz), f);
    fclose(f);
    TORCH_CHECK(rd == static_cast<size_t>(sz), "short read of ", path);
  }
  return parse_nlines_host_mt(text, false, n_threads);
}

// synthetic employee N-Triples FILE generator (bench tooling: writing
// 100M f-string lines from Python costs minutes; this streams the file
// in ~seconds with the same shape as parallel/synthetic.py)
int64_t gen_employee_nt_file(const std::string& path,
                             int64_t total_triples) {
  py::gil_scoped_release release;
  const int64_t n_emp = std::max<int64_t>(1, total_triples / 7);
  const int64_t n_dept = std::max<int64_t>(1, n_emp / 100);
  FILE* f = fopen(path.c_str(), "wb");
  TORCH_CHECK(f != nullptr, "cannot open ", path, " for writing");
  std::vector<char> buf(1 << 22);
  setvbuf(f, buf.data(), _IOFBF, buf.size());
  int64_t lines = 0;
  static const char* kPos[3] = {"Manager", "Developer", "Salesperson"};
  for (int64_t i = 0; i < n_emp; ++i) {
    long e = static_cast<long>(i);
    long d = static_cast<long>(i % n_dept);
    fprintf(f,
            "<http://synthetic/e%ld> <http://xmlns.com/foaf/0.1/name> "
            "\"name%ld\" .\n"
            "<http://synthetic/e%ld> "
            "<http://xmlns.com/foaf/0.1/workplaceHomepage> "
            "<http://synthetic/h%ld> .\n"
            "<http://synthetic/e%ld> "
            "<https://data.cityofchicago.org/resource/xzkq-xp2w/"
            "annual_salary> \"%ld\" .\n"
            "<http://synthetic/e%ld> "
            "<https://data.cityofchicago.org/resource/xzkq-xp2w/position> "
            "\"%s\" .\n"
            "<http://synthetic/e%ld> "
            "<https://data.cityofchicago.org/resource/xzkq-xp2w/email> "
            "\"e%ld@example.org\" .\n"
            "<http://synthetic/e%ld> "
            "<https://data.cityofchicago.org/resource/xzkq-xp2w/age> "
            "\"%ld\" .\n"
            "<http://synthetic/e%ld> "
            "<https://data.cityofchicago.org/resource/xzkq-xp2w/worksFor> "
            "<http://synthetic/d%ld> .\n",
            e, e, e, e, e, 30000 + (e * 37) % 120000, e,
            kPos[e % 3], e, e, e, 20 + (e * 13) % 50, e, d);
    lines += 7;
  }
  for (int64_t i = 0; i < n_dept; ++i) {
    long d = static_cast<long>(i);
    fprintf(f,
            "<http://synthetic/d%ld> "
            "<https://data.cityofchicago.org/resource/xzkq-xp2w/locatedIn> "
            "<http://synthetic/c%ld> .\n"
            "<http://synthetic/d%ld> "
            "<http://www.w3.org/2000/01/rdf-schema#label> \"dept %ld\" .\n",
            d, static_cast<long>(i % 1000), d, d);
    lines += 2;
  }
  fclose(f);
  return lines;
}

py::tuple parse_ntriples_host(const std::string& text) {
  return parse_nlines_host(text, false);
}

// N-Quads: optional 4th graph term; absent -> -1 in the id row
// (host bulk replacement for sparql_database.rs:1411 parse_nquads_and_add)
py::tuple parse_nquads_host(const std::string& text) {
  return parse_nlines_host(text, true);
}

// ------------------------------------------------------------ K4: group agg
// Hand-written GROUP BY hash aggregate (replaces reference
// execute_query.rs:404-475 aggregate_rows).
//
// One pass over (key, value) rows into an open-addressing global table,
// staged through a per-workgroup LDS table first: low-cardinality keys
// (the contended case) are pre-aggregated at LDS-atomic speed and flushed
// once per block, so global atomics see at most (#blocks x #groups)
// traffic; high-cardinality keys miss the small LDS table and go straight
// to the (uncontended) global table.  A compaction pass then emits the
// occupied slots.  MIN/MAX on doubles use the monotone bit encoding so
// unsigned 64-bit atomicMin/Max apply.
//
// Keys are caller-packed int64 (single var: (uint32)id; two vars:
// (a<<32)|b).  `empty_key` is a caller-chosen value no row can produce.
constexpr int kAggLdsSlots = 2048;   // 5 x 8 B x 2048 = 80 KiB LDS -> 2 WG/CU

__device__ __forceinline__ uint32_t h64(unsigned long long k) {
  k ^= k >> 33; k *= 0xff51afd7ed558ccdULL;
  k ^= k >> 33; k *= 0xc4ceb9fe1a85ec53ULL;
  k ^= k >> 33;
  return static_cast<uint32_t>(k);
}

__device__ __forceinline__ unsigned long long enc_double(double x) {
  unsigned long long b =
      static_cast<unsigned long long>(__double_as_longlong(x));
  return (b & 0x8000000000000000ULL) ? ~b : (b | 0x8000000000000000ULL);
}

__device__ __forceinline__ double dec_double(unsigned long long e) {
  unsigned long long b =
      (e & 0x8000000000000000ULL) ? (e & 0x7FFFFFFFFFFFFFFFULL) : ~e;
  return __longlong_as_double(static_cast<long long>(b));
}

// probe-run cap: at the <=50% load factor the host wrapper guarantees on
// the final attempt, expected linear-probe runs are ~2 slots; a run this
// long means the table is too small — flag overflow and let the host
// re-run bigger.  A full-table scan here (the naive failure mode) would
// cost minutes at millions of threads.
constexpr uint32_t kAggMaxProbe = 128;

__device__ __forceinline__ int64_t agg_claim_slot(
    unsigned long long* __restrict__ keys, uint32_t mask,
    unsigned long long key, unsigned long long empty,
    int32_t* __restrict__ overflow) {
  uint32_t h = h64(key) & mask;
  uint32_t cap = mask < kAggMaxProbe ? mask : kAggMaxProbe;
  for (uint32_t probe = 0; probe <= cap; ++probe) {
    unsigned long long cur = keys[h];
    if (cur == empty) {
      // resolve via CAS, not a re-read: atomics land in L2 and a stale
      // L1 line could keep reporting `empty` forever
      cur = atomicCAS(&keys[h], empty, key);
      if (cur == empty) return h;
    }
    if (cur == key) return h;
    h = (h + 1) & mask;
  }
  atomicExch(overflow, 1);  // table too loaded: caller re-runs bigger
  return -1;
}

__global__ void __launch_bounds__(kBlock)
group_agg_kernel(const int64_t* __restrict__ keys_in,
                 const double* __restrict__ vals, int64_t n,
                 unsigned long long empty,
                 unsigned long long* __restrict__ gkeys,
                 unsigned long long* __restrict__ gcnt,
                 double* __restrict__ gsum,
                 unsigned long long* __restrict__ gmn,
                 unsigned long long* __restrict__ gmx,
                 uint32_t gmask, int32_t* __restrict__ overflow,
                 int use_lds) {
  __shared__ unsigned long long lkeys[kAggLdsSlots];
  __shared__ unsigned long long lcnt[kAggLdsSlots];
  __shared__ double lsum[kAggLdsSlots];
  __shared__ unsigned long long lmn[kAggLdsSlots];
  __shared__ unsigned long long lmx[kAggLdsSlots];
  for (int i = threadIdx.x; i < kAggLdsSlots; i += blockDim.x) {
    lkeys[i] = empty;
    lcnt[i] = 0ULL;
    lsum[i] = 0.0;
    lmn[i] = ~0ULL;
    lmx[i] = 0ULL;
  }
  __syncthreads();
  const bool want_val = vals != nullptr;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if ((i & 0x3FF) == 0 && *overflow) return;  // abort a doomed attempt
    unsigned long long key = static_cast<unsigned long long>(keys_in[i]);
    double v = want_val ? vals[i] : 0.0;
    unsigned long long ev = want_val ? enc_double(v) : 0ULL;
    // LDS attempt: short probe run, then spill to the global table.
    // use_lds=0 (high estimated cardinality: most keys would miss the
    // small LDS table anyway) goes straight to the global table.
    uint32_t h = h64(key) & (kAggLdsSlots - 1);
    int64_t slot = -1;
    for (int p = 0; use_lds && p < 8; ++p) {
      unsigned long long cur = lkeys[h];
      if (cur == key) { slot = h; break; }
      if (cur == empty) {
        unsigned long long prev = atomicCAS(&lkeys[h], empty, key);
        if (prev == empty || prev == key) { slot = h; break; }
        continue;  // re-read: someone else claimed it this cycle
      }
      h = (h + 1) & (kAggLdsSlots - 1);
    }
    if (slot >= 0) {
      atomicAdd(&lcnt[slot], 1ULL);
      if (want_val) {
        if (gsum) atomicAdd(&lsum[slot], v);
        if (gmn) atomicMin(&lmn[slot], ev);
        if (gmx) atomicMax(&lmx[slot], ev);
      }
    } else {
      int64_t gs = agg_claim_slot(gkeys, gmask, key, empty, overflow);
      if (gs < 0) return;
      atomicAdd(&gcnt[gs], 1ULL);
      if (want_val) {
        if (gsum) atomicAdd(&gsum[gs], v);
        if (gmn) atomicMin(&gmn[gs], ev);
        if (gmx) atomicMax(&gmx[gs], ev);
      }
    }
  }
  __syncthreads();
  // flush the block's LDS pre-aggregates into the global table
  for (int i = threadIdx.x; i < kAggLdsSlots; i += blockDim.x) {
    if (lkeys[i] == empty) continue;
    int64_t gs = agg_claim_slot(gkeys, gmask, lkeys[i], empty, overflow);
    if (gs < 0) return;
    atomicAdd(&gcnt[gs], lcnt[i]);
    if (gsum) atomicAdd(&gsum[gs], lsum[i]);
    if (gmn) atomicMin(&gmn[gs], lmn[i]);
    if (gmx) atomicMax(&gmx[gs], lmx[i]);
  }
}

__global__ void agg_compact(const unsigned long long* __restrict__ gkeys,
                            const unsigned long long* __restrict__ gcnt,
                            const double* __restrict__ gsum,
                            const unsigned long long* __restrict__ gmn,
                            const unsigned long long* __restrict__ gmx,
                            int64_t cap, unsigned long long empty,
                            int64_t* __restrict__ out_keys,
                            int64_t* __restrict__ out_cnt,
                            double* __restrict__ out_sum,
                            double* __restrict__ out_mn,
                            double* __restrict__ out_mx,
                            int32_t* __restrict__ counter) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < cap;
       i += (int64_t)gridDim.x * blockDim.x) {
    unsigned long long k = gkeys[i];
    if (k == empty) continue;
    int32_t idx = atomicAdd(counter, 1);
    out_keys[idx] = static_cast<int64_t>(k);
    out_cnt[idx] = static_cast<int64_t>(gcnt[i]);
    if (out_sum) out_sum[idx] = gsum[i];
    if (out_mn) out_mn[idx] = dec_double(gmn[i]);
    if (out_mx) out_mx[idx] = dec_double(gmx[i]);
  }
}

// host wrapper: (keys[n] int64, vals[n] double|None) ->
//   (keys[g], cnt[g], sum[g]|None, min[g]|None, max[g]|None)
py::tuple group_aggregate(at::Tensor keys, c10::optional<at::Tensor> vals,
                          bool want_sum, bool want_min, bool want_max,
                          int64_t empty_key, int64_t groups_hint) {
  TORCH_CHECK(keys.is_cuda() && keys.dtype() == at::kLong && keys.dim() == 1);
  const int64_t n = keys.numel();
  const double* vptr = nullptr;
  if (vals.has_value()) {
    TORCH_CHECK(vals->is_cuda() && vals->dtype() == at::kDouble
                && vals->numel() == n);
    vptr = vals->data_ptr<double>();
  }
  const bool want_val = vptr != nullptr;
  auto opts_i64 = keys.options();
  auto opts_f64 = keys.options().dtype(at::kDouble);
  auto opts_i32 = keys.options().dtype(at::kInt);
  auto stream = cur_stream();
  const unsigned long long empty =
      static_cast<unsigned long long>(empty_key);
  // size from the caller's sampled cardinality estimate (4x the upper
  // bound keeps load <=25%) or default to ~n/4; retry x8 on overflow —
  // the final rung 2*n can never overflow.  A failed attempt aborts
  // early (probe cap + overflow poll), costing ~one extra key pass.
  int64_t cap = 1 << 14;
  if (groups_hint > 0) {
    while (cap < 4 * groups_hint && cap < 2 * n) cap <<= 1;
  } else {
    while (cap < n / 4 && cap < (1LL << 27)) cap <<= 1;
  }
  const bool use_lds =
      groups_hint > 0 ? (groups_hint <= 3000) : true;
  for (;;) {
    auto gkeys = at::full({cap}, empty_key, opts_i64);
    auto gcnt = at::zeros({cap}, opts_i64);
    at::Tensor gsum, gmn, gmx;
    double* gsum_p = nullptr;
    unsigned long long* gmn_p = nullptr;
    unsigned long long* gmx_p = nullptr;
    if (want_val && want_sum) {
      gsum = at::zeros({cap}, opts_f64);
      gsum_p = gsum.data_ptr<double>();
    }
    if (want_val && want_min) {
      gmn = at::full({cap}, -1, opts_i64);  // ~0ULL = +inf in the encoding
      gmn_p = reinterpret_cast<unsigned long long*>(gmn.data_ptr<int64_t>());
    }
    if (want_val && want_max) {
      gmx = at::zeros({cap}, opts_i64);     // 0 = -inf in the encoding
      gmx_p = reinterpret_cast<unsigned long long*>(gmx.data_ptr<int64_t>());
    }
    auto overflow = at::zeros({1}, opts_i32);
    if (n > 0) {
      hipLaunchKernelGGL(group_agg_kernel, dim3(grid_for(n)), dim3(kBlock), 0,
                         stream, keys.data_ptr<int64_t>(), vptr, n, empty,
                         reinterpret_cast<unsigned long long*>(
                             gkeys.data_ptr<int64_t>()),
                         reinterpret_cast<unsigned long long*>(
                             gcnt.data_ptr<int64_t>()),
                         gsum_p, gmn_p, gmx_p,
                         static_cast<uint32_t>(cap - 1),
                         overflow.data_ptr<int32_t>(),
                         use_lds ? 1 : 0);
      HIP_OK(hipGetLastError());
    }
    if (n > 0 && overflow.item<int32_t>() != 0) {
      int64_t full = 1;
      while (full < 2 * n) full <<= 1;
      TORCH_CHECK(cap < full, "group_aggregate: overflow at 2*n capacity");
      cap = std::min(cap << 3, full);
      continue;
    }
    auto out_keys = at::empty({cap}, opts_i64);
    auto out_cnt = at::empty({cap}, opts_i64);
    at::Tensor out_sum, out_mn, out_mx;
    double* os = nullptr; double* omn = nullptr; double* omx = nullptr;
    if (gsum_p) { out_sum = at::empty({cap}, opts_f64);
                  os = out_sum.data_ptr<double>(); }
    if (gmn_p) { out_mn = at::empty({cap}, opts_f64);
                 omn = out_mn.data_ptr<double>(); }
    if (gmx_p) { out_mx = at::empty({cap}, opts_f64);
                 omx = out_mx.data_ptr<double>(); }
    auto counter = at::zeros({1}, opts_i32);
    hipLaunchKernelGGL(agg_compact, dim3(grid_for(cap)), dim3(kBlock), 0,
                       stream,
                       reinterpret_cast<unsigned long long*>(
                           gkeys.data_ptr<int64_t>()),
                       reinterpret_cast<unsigned long long*>(
                           gcnt.data_ptr<int64_t>()),
                       gsum_p, gmn_p, gmx_p, cap, empty,
                       out_keys.data_ptr<int64_t>(),
                       out_cnt.data_ptr<int64_t>(), os, omn, omx,
                       counter.data_ptr<int32_t>());
    HIP_OK(hipGetLastError());
    int64_t g = counter.item<int32_t>();
    py::tuple t(5);
    t[0] = out_keys.narrow(0, 0, g);
    t[1] = out_cnt.narrow(0, 0, g);
    t[2] = gsum_p ? py::cast(out_sum.narrow(0, 0, g)) : py::none();
    t[3] = gmn_p ? py::cast(out_mn.narrow(0, 0, g)) : py::none();
    t[4] = gmx_p ? py::cast(out_mx.narrow(0, 0, g)) : py::none();
    return t;
  }
}

// ------------------------------------------------- K6: small-N fixpoint
// Persistent SINGLE-WORKGROUP semi-naive fixpoint for small working sets
// (the deep-taxonomy shape = 10 000 rounds of 1-fact
// deltas, where per-round kernel-launch + host-sync overhead dominates
// any columnar formulation).  One 1024-thread block runs the WHOLE
// fixpoint: rounds are __syncthreads() boundaries (ns, not µs), and all
// inter-thread data passes stay inside one CU so workgroup-scope fences
// suffice (MI355X guide §Workgroup dispatch: no agent-scope traffic).
//
// Rule language (host driver rejects anything else and falls back):
//   kind=1 (join):  (a P1 b), (c P2 d) -> conclusions, exactly ONE shared
//                   var between the premises, predicates constant;
//   kind=0 (copy):  (a P1 b) -> conclusions.
// Conclusions (<=2 per rule): (src_s Pc src_o), srcs in {p1.s,p1.o,p2.s,
// p2.o}.  Covers transitive closure, type propagation (deep taxonomy),
// ancestor programs.
//
// State per predicate (global memory, host-allocated):
//   - pair dedup set: open addressing on packed (s<<32)|o u64, CAS claim;
//   - optional adjacency by-s / by-o: chained hash (heads + append-only
//     {key,val,next} entries), only for directions some rule probes.
// The fact log `out` doubles as the delta queue: [d_lo, d_hi) is the
// current round's delta; appended facts form the next round's.
//
// Ref semantics: datalog semi_naive.rs:17-86 (one-shared-var case).
constexpr int kFxBlock = 1024;
constexpr unsigned long long kFxEmpty = ~0ULL;  // (UNBOUND,UNBOUND) pair

struct FxRule {
  int32_t kind;      // 0 copy, 1 join
  int32_t p1, j1;    // premise1 pred idx; shared-var position (0=s,1=o)
  int32_t p2, j2;    // premise2 (join only)
  int32_t n_conc;
  int32_t c_pred[2];
  int32_t c_s_src[2];  // 0=p1.s 1=p1.o 2=p2.s 3=p2.o
  int32_t c_o_src[2];
};

struct FxPred {
  unsigned long long* set;   // pair dedup table
  uint32_t set_mask;
  int32_t* adj_s_heads; uint32_t adj_s_mask;   // by-subject chains
  int32_t* adj_o_heads; uint32_t adj_o_mask;   // by-object chains
};

struct FxState {
  // shared append-only pools (all predicates)
  int64_t* adj_entries;      // packed (key<<32)|val
  int32_t* adj_next;
  int32_t* adj_pred_dir;     // unused slot kept for debug
  int32_t* out_s; int32_t* out_p; int32_t* out_o;  // fact log (pred idx)
  int32_t* counters;         // [0]=out_n [1]=adj_n [2]=overflow [3]=rounds
  int64_t out_cap, adj_cap;
};

__device__ __forceinline__ bool fx_set_insert(const FxPred& pr,
                                              uint32_t s, uint32_t o,
                                              int32_t* overflow) {
  unsigned long long key =
      (static_cast<unsigned long long>(s) << 32) | o;
  uint32_t h = h64(key) & pr.set_mask;
  uint32_t cap = pr.set_mask < kAggMaxProbe ? pr.set_mask : kAggMaxProbe;
  for (uint32_t p = 0; p <= cap; ++p) {
    unsigned long long cur = pr.set[h];
    if (cur == kFxEmpty) {
      cur = atomicCAS(&pr.set[h], kFxEmpty, key);  // L2 truth, not L1
      if (cur == kFxEmpty) return true;
    }
    if (cur == key) return false;
    h = (h + 1) & pr.set_mask;
  }
  atomicExch(overflow, 1);
  return false;
}

// L1-bypassing relaxed loads: atomic publishes (atomicExch heads,
// atomicAdd counters) land in L2 and never refresh even this CU's own
// L1, so every read of a mutable shared structure goes L2-served.
__device__ __forceinline__ int32_t fx_ld(const int32_t* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}
__device__ __forceinline__ int64_t fx_ld64(const int64_t* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

// Allocation counters / overflow flag / optional LDS delta ring.  In the
// persistent single-WG kernel these live in LDS (per-emit bookkeeping at
// LDS-atomic speed instead of ~4 L2 round trips); the multi-block seed
// kernel passes the global counters instead.
constexpr int kFxRing = 4096;

struct FxCtx {
  int32_t* out_n;
  int32_t* adj_n;
  int32_t* overflow;
  int32_t* ring_s; int32_t* ring_p; int32_t* ring_o;  // LDS or null
  int32_t ring_base;   // fact-log index of ring slot 0 (this round)
};

__device__ __forceinline__ void fx_adj_insert(int32_t* heads, uint32_t mask,
                                              uint32_t key, uint32_t val,
                                              FxCtx& cx, FxState& st) {
  int32_t e = atomicAdd(cx.adj_n, 1);
  if (e >= st.adj_cap) { atomicExch(cx.overflow, 1); return; }
  __hip_atomic_store(
      &st.adj_entries[e],
      static_cast<int64_t>((static_cast<unsigned long long>(key) << 32) | val),
      __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  // lock-free push: next[e] must hold the correct predecessor BEFORE the
  // head CAS makes e reachable, or a concurrent walker would read junk
  uint32_t h = h32(key) & mask;
  int32_t prev = fx_ld(&heads[h]);
  for (;;) {
    __hip_atomic_store(&st.adj_next[e], prev, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_AGENT);
    __threadfence_block();
    int32_t seen = atomicCAS(&heads[h], prev, e);
    if (seen == prev) break;
    prev = seen;
  }
}

// emit one derived fact: dedup -> fact log (= next delta) -> adjacencies
__device__ __forceinline__ void fx_emit(int32_t pidx, uint32_t s, uint32_t o,
                                        const FxPred* preds, FxCtx& cx,
                                        FxState& st) {
  const FxPred& pr = preds[pidx];
  if (!fx_set_insert(pr, s, o, cx.overflow)) return;
  int32_t idx = atomicAdd(cx.out_n, 1);
  if (idx >= st.out_cap) { atomicExch(cx.overflow, 1); return; }
  __hip_atomic_store(&st.out_s[idx], static_cast<int32_t>(s),
                     __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  __hip_atomic_store(&st.out_p[idx], pidx,
                     __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  __hip_atomic_store(&st.out_o[idx], static_cast<int32_t>(o),
                     __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  if (cx.ring_s) {
    int32_t r = idx - cx.ring_base;
    if (r >= 0 && r < kFxRing) {
      cx.ring_s[r] = static_cast<int32_t>(s);
      cx.ring_p[r] = pidx;
      cx.ring_o[r] = static_cast<int32_t>(o);
    }
  }
  if (pr.adj_s_heads)
    fx_adj_insert(pr.adj_s_heads, pr.adj_s_mask, s, o, cx, st);
  if (pr.adj_o_heads)
    fx_adj_insert(pr.adj_o_heads, pr.adj_o_mask, o, s, cx, st);
}

__device__ __forceinline__ uint32_t fx_src(int src, uint32_t s1, uint32_t o1,
                                           uint32_t s2, uint32_t o2) {
  switch (src) {
    case 0: return s1;
    case 1: return o1;
    case 2: return s2;
    default: return o2;
  }
}

// compile-time caps so ALL rule/dispatch metadata fits in LDS (a tiny
// few KB): per-round reads of it then cost LDS cycles, not L2 latency —
// at 10 000 one-fact rounds the metadata reads would otherwise dominate
constexpr int kFxMaxRules = 64;
constexpr int kFxMaxPreds = 16;
constexpr int kFxMaxDisp = 128;

__global__ void __launch_bounds__(kFxBlock)
fx_kernel(const FxRule* __restrict__ g_rules, int n_rules,
          const FxPred* __restrict__ g_preds, int n_preds,
          const int32_t* __restrict__ g_disp_off,   // [n_preds+1]
          const int32_t* __restrict__ g_disp_rule,  // rule index
          const int32_t* __restrict__ g_disp_side,  // 0: fact is premise1
          FxState st, int64_t n_init, int64_t adj_seeded,
          int64_t max_rounds) {
  __shared__ int32_t s_dlo, s_dhi, s_use_ring;
  __shared__ int32_t sh_out_n, sh_adj_n, sh_overflow;
  __shared__ FxRule rules[kFxMaxRules];
  __shared__ FxPred preds[kFxMaxPreds];
  __shared__ int32_t disp_off[kFxMaxPreds + 1];
  __shared__ int32_t disp_rule[kFxMaxDisp];
  __shared__ int32_t disp_side[kFxMaxDisp];
  // double-buffered: this round's emits write parity buffer `round & 1`
  // while readers consume the previous round's buffer
  __shared__ int32_t ring_s[2][kFxRing], ring_p[2][kFxRing],
      ring_o[2][kFxRing];
  if (threadIdx.x == 0) {
    s_dlo = 0;
    s_dhi = static_cast<int32_t>(n_init);
    s_use_ring = 0;
    sh_out_n = static_cast<int32_t>(n_init);
    sh_adj_n = static_cast<int32_t>(adj_seeded);
    sh_overflow = 0;
    for (int i = 0; i < n_rules; ++i) rules[i] = g_rules[i];
    for (int i = 0; i < n_preds; ++i) preds[i] = g_preds[i];
    int nd = g_disp_off[n_preds];
    for (int i = 0; i <= n_preds; ++i) disp_off[i] = g_disp_off[i];
    for (int i = 0; i < nd; ++i) {
      disp_rule[i] = g_disp_rule[i];
      disp_side[i] = g_disp_side[i];
    }
  }
  __syncthreads();
  int64_t rounds_done = 0;
  for (int64_t round = 0; round < max_rounds; ++round) {
    int32_t dlo = s_dlo, dhi = s_dhi;
    if (dlo >= dhi) break;
    bool ring_ok = s_use_ring != 0;
    const int wb = static_cast<int>(round & 1), rb = wb ^ 1;
    FxCtx cx{&sh_out_n, &sh_adj_n, &sh_overflow,
             ring_s[wb], ring_p[wb], ring_o[wb], dhi};
    for (int32_t i = dlo + threadIdx.x; i < dhi; i += blockDim.x) {
      uint32_t fs, fo; int32_t fp;
      if (ring_ok) {
        fs = static_cast<uint32_t>(ring_s[rb][i - dlo]);
        fp = ring_p[rb][i - dlo];
        fo = static_cast<uint32_t>(ring_o[rb][i - dlo]);
      } else {
        fs = static_cast<uint32_t>(fx_ld(&st.out_s[i]));
        fp = fx_ld(&st.out_p[i]);
        fo = static_cast<uint32_t>(fx_ld(&st.out_o[i]));
      }
      for (int32_t d = disp_off[fp]; d < disp_off[fp + 1]; ++d) {
        const FxRule& r = rules[disp_rule[d]];
        int side = disp_side[d];
        if (r.kind == 0) {
          for (int c = 0; c < r.n_conc; ++c)
            fx_emit(r.c_pred[c], fx_src(r.c_s_src[c], fs, fo, 0, 0),
                    fx_src(r.c_o_src[c], fs, fo, 0, 0), preds, cx, st);
          continue;
        }
        // join: this fact binds one premise; probe the other's adjacency
        int other_p = side == 0 ? r.p2 : r.p1;
        int my_j = side == 0 ? r.j1 : r.j2;
        int ot_j = side == 0 ? r.j2 : r.j1;
        uint32_t v = my_j == 0 ? fs : fo;   // shared-var value
        const FxPred& op = preds[other_p];
        int32_t* heads = ot_j == 0 ? op.adj_s_heads : op.adj_o_heads;
        uint32_t mask = ot_j == 0 ? op.adj_s_mask : op.adj_o_mask;
        if (!heads) continue;  // driver guarantees presence; safety
        uint32_t h = h32(v) & mask;
        for (int32_t e = fx_ld(&heads[h]); e >= 0;
             e = fx_ld(&st.adj_next[e])) {
          unsigned long long kv =
              static_cast<unsigned long long>(fx_ld64(&st.adj_entries[e]));
          if (static_cast<uint32_t>(kv >> 32) != v) continue;
          uint32_t w = static_cast<uint32_t>(kv);
          uint32_t s2, o2;
          if (ot_j == 0) { s2 = v; o2 = w; } else { s2 = w; o2 = v; }
          uint32_t s1, o1;
          if (side == 0) { s1 = fs; o1 = fo; }
          else { s1 = s2; o1 = o2; s2 = fs; o2 = fo; }
          // (when side==1 the roles swap: this fact IS premise2)
          for (int c = 0; c < r.n_conc; ++c)
            fx_emit(r.c_pred[c], fx_src(r.c_s_src[c], s1, o1, s2, o2),
                    fx_src(r.c_o_src[c], s1, o1, s2, o2), preds, cx, st);
        }
      }
    }
    __threadfence_block();
    __syncthreads();
    if (threadIdx.x == 0) {
      s_dlo = dhi;
      int32_t n = sh_out_n;
      if (n > st.out_cap) n = static_cast<int32_t>(st.out_cap);
      s_dhi = n;
      // next round may read its delta from the LDS ring only if EVERY
      // fact appended this round landed in it
      s_use_ring = (n - dhi) <= kFxRing ? 1 : 0;
    }
    __syncthreads();
    rounds_done = round + 1;
    if (sh_overflow) break;  // overflow: host falls back
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    st.counters[0] = sh_out_n;
    st.counters[1] = sh_adj_n;
    st.counters[2] = sh_overflow;
    st.counters[3] = static_cast<int32_t>(rounds_done);
  }
}

// seed kernel: parallel insert of the initial facts (dedup + adjacency)
__global__ void fx_seed(const int32_t* __restrict__ s,
                        const int32_t* __restrict__ p,
                        const int32_t* __restrict__ o, int64_t n,
                        const FxPred* __restrict__ preds, FxState st) {
  FxCtx cx{&st.counters[0], &st.counters[1], &st.counters[2],
           nullptr, nullptr, nullptr, 0};
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    fx_emit(p[i], static_cast<uint32_t>(s[i]), static_cast<uint32_t>(o[i]),
            preds, cx, st);
  }
}

// host wrapper.  rules_flat: [n_rules,12] int32 rows =
//   kind,p1,j1,p2,j2,n_conc,c0_pred,c0_s,c0_o,c1_pred,c1_s,c1_o
// adj_need: [n_preds,2] bool-ish int32 (by_s, by_o)
// facts: int32 columns with p already mapped to predicate indexes
py::tuple small_fixpoint(at::Tensor rules_flat, at::Tensor adj_need,
                         at::Tensor disp_off, at::Tensor disp_rule,
                         at::Tensor disp_side,
                         at::Tensor fs, at::Tensor fp, at::Tensor fo,
                         int64_t budget, int64_t max_rounds) {
  TORCH_CHECK(fs.is_cuda() && fs.dtype() == at::kInt);
  const int64_t n_init = fs.numel();
  const int n_rules = rules_flat.size(0);
  const int n_preds = adj_need.size(0);
  auto opts_i32 = fs.options();
  auto opts_i64 = fs.options().dtype(at::kLong);
  auto stream = cur_stream();

  // ---- host-side structure assembly (tiny) ----
  auto rules_h = rules_flat.to(at::kCPU).contiguous();
  auto need_h = adj_need.to(at::kCPU).contiguous();
  std::vector<FxRule> rules_v(n_rules);
  const int32_t* rr = rules_h.data_ptr<int32_t>();
  for (int i = 0; i < n_rules; ++i) {
    const int32_t* q = rr + i * 12;
    FxRule& r = rules_v[i];
    r.kind = q[0]; r.p1 = q[1]; r.j1 = q[2]; r.p2 = q[3]; r.j2 = q[4];
    r.n_conc = q[5];
    r.c_pred[0] = q[6]; r.c_s_src[0] = q[7]; r.c_o_src[0] = q[8];
    r.c_pred[1] = q[9]; r.c_s_src[1] = q[10]; r.c_o_src[1] = q[11];
  }
  int64_t set_cap = 64;
  while (set_cap < 2 * (n_init + budget)) set_cap <<= 1;
  int64_t head_cap = 64;
  while (head_cap < n_init + budget) head_cap <<= 1;

  std::vector<at::Tensor> keepalive;
  std::vector<FxPred> preds_v(n_preds);
  const int32_t* nd = need_h.data_ptr<int32_t>();
  for (int i = 0; i < n_preds; ++i) {
    auto set_t = at::full({set_cap}, -1, opts_i64);
    keepalive.push_back(set_t);
    preds_v[i].set = reinterpret_cast<unsigned long long*>(
        set_t.data_ptr<int64_t>());
    preds_v[i].set_mask = static_cast<uint32_t>(set_cap - 1);
    preds_v[i].adj_s_heads = nullptr;
    preds_v[i].adj_o_heads = nullptr;
    preds_v[i].adj_s_mask = preds_v[i].adj_o_mask = 0;
    if (nd[i * 2 + 0]) {
      auto h = at::full({head_cap}, -1, opts_i32);
      keepalive.push_back(h);
      preds_v[i].adj_s_heads = h.data_ptr<int32_t>();
      preds_v[i].adj_s_mask = static_cast<uint32_t>(head_cap - 1);
    }
    if (nd[i * 2 + 1]) {
      auto h = at::full({head_cap}, -1, opts_i32);
      keepalive.push_back(h);
      preds_v[i].adj_o_heads = h.data_ptr<int32_t>();
      preds_v[i].adj_o_mask = static_cast<uint32_t>(head_cap - 1);
    }
  }
  const int64_t out_cap = n_init + budget;
  const int64_t adj_cap = 2 * out_cap;
  auto out_s = at::empty({out_cap}, opts_i32);
  auto out_p = at::empty({out_cap}, opts_i32);
  auto out_o = at::empty({out_cap}, opts_i32);
  auto adj_e = at::empty({adj_cap}, opts_i64);
  auto adj_n = at::empty({adj_cap}, opts_i32);
  auto counters = at::zeros({8}, opts_i32);
  FxState st;
  st.adj_entries = adj_e.data_ptr<int64_t>();
  st.adj_next = adj_n.data_ptr<int32_t>();
  st.adj_pred_dir = nullptr;
  st.out_s = out_s.data_ptr<int32_t>();
  st.out_p = out_p.data_ptr<int32_t>();
  st.out_o = out_o.data_ptr<int32_t>();
  st.counters = counters.data_ptr<int32_t>();
  st.out_cap = out_cap;
  st.adj_cap = adj_cap;

  // device copies of the rule/pred tables
  auto rules_bytes = at::from_blob(rules_v.data(),
      {static_cast<int64_t>(n_rules * sizeof(FxRule))},
      at::TensorOptions().dtype(at::kByte)).clone().to(fs.device());
  auto preds_bytes = at::from_blob(preds_v.data(),
      {static_cast<int64_t>(n_preds * sizeof(FxPred))},
      at::TensorOptions().dtype(at::kByte)).clone().to(fs.device());
  keepalive.push_back(rules_bytes);
  keepalive.push_back(preds_bytes);
  auto d_off = disp_off.to(fs.device(), at::kInt).contiguous();
  auto d_rule = disp_rule.to(fs.device(), at::kInt).contiguous();
  auto d_side = disp_side.to(fs.device(), at::kInt).contiguous();

  const FxPred* dp = reinterpret_cast<const FxPred*>(
      preds_bytes.data_ptr<uint8_t>());
  if (n_init > 0) {
    hipLaunchKernelGGL(fx_seed, dim3(grid_for(n_init)), dim3(kBlock), 0,
                       stream, fs.data_ptr<int32_t>(), fp.data_ptr<int32_t>(),
                       fo.data_ptr<int32_t>(), n_init, dp, st);
    HIP_OK(hipGetLastError());
  }
  // NOTE: the seed pass already appended the init facts to the log; the
  // fixpoint's first delta is [0, counters[0]) == all seeded facts.
  auto seed_cts = counters.to(at::kCPU);
  int64_t seeded = seed_cts[0].item<int32_t>();
  int64_t adj_seeded = seed_cts[1].item<int32_t>();
  TORCH_CHECK(n_rules <= kFxMaxRules && n_preds <= kFxMaxPreds
              && d_rule.numel() <= kFxMaxDisp,
              "small_fixpoint: program exceeds LDS metadata caps");
  hipLaunchKernelGGL(fx_kernel, dim3(1), dim3(kFxBlock), 0, stream,
                     reinterpret_cast<const FxRule*>(
                         rules_bytes.data_ptr<uint8_t>()),
                     n_rules, dp, n_preds, d_off.data_ptr<int32_t>(),
                     d_rule.data_ptr<int32_t>(), d_side.data_ptr<int32_t>(),
                     st, seeded, adj_seeded, max_rounds);
  HIP_OK(hipGetLastError());
  auto cts = counters.to(at::kCPU);
  int32_t n_out = cts[0].item<int32_t>();
  int32_t overflow = cts[2].item<int32_t>();
  int32_t rounds = cts[3].item<int32_t>();
  if (n_out > out_cap) n_out = static_cast<int32_t>(out_cap);
  py::tuple t(6);
  t[0] = out_s.narrow(0, 0, n_out);
  t[1] = out_p.narrow(0, 0, n_out);
  t[2] = out_o.narrow(0, 0, n_out);
  t[3] = py::int_(seeded);
  t[4] = py::int_(overflow);
  t[5] = py::int_(rounds);
  return t;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("small_fixpoint", &small_fixpoint,
        "K6 persistent single-workgroup semi-naive fixpoint for small "
        "working sets (deep-taxonomy class) -> (s, p_idx, o, n_seeded, "
        "overflow, rounds)");
  m.def("group_aggregate", &group_aggregate,
        "K4 LDS-staged hash GROUP BY: (keys, vals?, sum, min, max, empty, "
        "groups_hint) -> (group keys, counts, sums?, mins?, maxs?)");
  m.def("parse_ntriples_host", &parse_ntriples_host,
        "bulk N-Triples parse -> (local-id triples, strings, fallback lines)");
  m.def("parse_nquads_host", &parse_nquads_host,
        "bulk N-Quads parse -> (local-id quads, strings, fallback lines)");
  m.def("parse_ntriples_host_mt", &parse_ntriples_host_mt,
        "multithreaded bulk N-Triples parse (chunk-per-thread + merged "
        "intern tables; GIL released)");
  m.def("parse_nquads_host_mt", &parse_nquads_host_mt,
        "multithreaded bulk N-Quads parse");
  m.def("parse_ntriples_file_mt", &parse_ntriples_file_mt,
        "read + multithreaded-parse an N-Triples file");
  m.def("parse_ntriples_file_encode", &parse_ntriples_file_encode,
        "file -> parallel parse -> intern into the Python dictionary -> "
        "global-id int32 rows");
  m.def("gen_employee_nt_file", &gen_employee_nt_file,
        "write a synthetic employee N-Triples file (bench tooling)");
  m.def("vocab_create", &vocab_create, "native bulk-vocabulary annex");
  m.def("vocab_len", &vocab_len);
  m.def("vocab_lookup", &vocab_lookup);
  m.def("vocab_insert", &vocab_insert);
  m.def("vocab_get", &vocab_get);
  m.def("vocab_decode_batch", &vocab_decode_batch);
  m.def("vocab_value", &vocab_value);
  m.def("vocab_values", &vocab_values);
  m.def("vocab_export_strings", &vocab_export_strings);
  m.def("stats_gather", &stats_gather, "K8 one-pass database statistics");
  m.def("vocab_create_seeded", &vocab_create_seeded,
        "create a vocab annex pre-seeded with the frozen Python prefix");
  m.def("parse_ntriples_file_annex", &parse_ntriples_file_annex,
        "file -> parallel parse -> intern into the native vocab annex");
  m.doc() = "kolibrie_amd native CDNA4 kernels (gfx950)";
  m.def("probe_exact", &probe_exact,
        "K1 scan-probe, packed (a,b) exact keys -> (li, b, z)");
  m.def("probe_range", &probe_range,
        "K1 scan-probe, leading-component range -> (li, b, z)");
  m.def("hash_join", &hash_join,
        "K2 chained hash join over int32 key columns -> (li, ri)");
  m.def("chain_tile", []() { return static_cast<int64_t>(kTile); },
        "seed rows per chain tile (win buffer sizing)");
  m.def("chain_count", &chain_count,
        "fused COUNT(*) over a seed scan + probe-hop chain",
        py::arg("seed_b"), py::arg("seed_z"), py::arg("hop_key32"),
        py::arg("hop_src"), py::arg("hop_table"),
        py::arg("hop_dmin") = std::vector<int64_t>{});
  m.def("chain_count_into", &chain_count_into,
        "allocation/sync-free chain count into caller buffers "
        "(hipGraph-capturable)");
  m.def("register_chain_serve", &register_chain_serve,
        "cache a COUNT chain's launch arguments C++-side; returns handle",
        py::arg("seed_b"), py::arg("seed_z"), py::arg("hop_key32"),
        py::arg("hop_src"), py::arg("hop_table"),
        py::arg("hop_dmin") = std::vector<int64_t>{});
  m.def("serve_chain_count", &serve_chain_count,
        py::call_guard<py::gil_scoped_release>(),
        "one-call serving: direct launches + pinned 8-byte readback");
  m.def("release_chain_serve", &release_chain_serve,
        "drop a cached serve handle (store changed)");
  m.def("build_count_table32", &build_count_table32,
        "packed-u32 (val<<7)|count open-addressing table (half the L2 "
        "footprint; requires val < 2^25, count < 128)");
  m.def("build_count_table", &build_count_table,
        "open-addressing (value -> match count) table from packed "
        "(val<<32)|count entries, for hashed chain-count hops");
  m.def("probe_fused", &probe_fused,
        "K1 fused probe: inline key pack + merge-path + carry emit");
  m.def("probe_exact_counts", &probe_exact_counts,
        "K1 count-only exact probe -> per-row match counts");
  m.def("probe_range_counts", &probe_range_counts,
        "K1 count-only range probe -> per-row match counts");
  m.def("hash_join_counts", &hash_join_counts,
        "K2 count-only hash join -> per-left-row match counts");
  m.def("filter_bytecode", &filter_bytecode,
        "K5 filter bytecode evaluation -> bool mask");
}
