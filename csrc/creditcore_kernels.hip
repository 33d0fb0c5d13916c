// creditcore HIP kernels — gfx950 (MI355X) native scoring hot path.
//
// Replaces the reference's CPU hot loops (sklearn RF predict_proba,
// alibi-detect IForest.predict and TabularDrift.predict, invoked per request
// at /root/reference/databricks/src/02-register-model.ipynb cell-9 via
// app/main.py:72) with CDNA4 kernels over the flat buffers produced by
// creditcore/pack.py:
//
//   forest_kernel    — node-SoA BFS forest traversal (classifier + iforest).
//                      One thread per (row, tree-chunk); per-row features are
//                      staged in LDS column-major so the divergent, dynamically
//                      indexed feature lookups hit LDS instead of scratch.
//                      Leaf sums accumulate in f64 (one atomicAdd per thread)
//                      for bit-stable parity with the f64 CPU reference.
//   finalize_kernel  — P(default) = acc/n_trees; isolation-forest anomaly
//                      score 2^(-depth/denom) + offset and outlier threshold.
//   cat_hist_kernel  — per-categorical-feature batch histograms (LDS-partial,
//                      one global atomicAdd per non-zero bin per block).
//   ks_kernel        — exact two-sample K-S D per numeric feature: bitonic
//                      sort of the batch column in LDS, then per-element
//                      binary search into the sorted reference (L2-resident)
//                      evaluating |F_ref - F_batch| at both one-sided limits.
//
// Design notes (MI355X): wavefront = 64; block = 256 (4 waves); grids are
// sized ≥ ~2048 blocks where the batch allows so all 256 CUs across the
// 8 XCDs see work; everything stays on the caller's HIP stream (the engine's
// private per-replica stream) — no host sync inside.

#include <torch/extension.h>

#include <c10/hip/HIPGuard.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#define BLOCK 256
#define N_CAT 9
#define N_NUM 14
#define MAX_DRIFT_ROWS 16384  // LDS cap for the K-S sort (64 KiB of f32)
static constexpr size_t KS_LDS_BYTES_MAX = 160 * 1024 - 1024;  // gfx950 LDS/CU

#define HIP_CHECK(expr)                                              \
  do {                                                               \
    hipError_t _e = (expr);                                          \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e)); \
  } while (0)

// ---------------------------------------------------------------------------
// Forest traversal
// ---------------------------------------------------------------------------

// DIRECT=false: classifier — node.feat is a virtual feature resolved through
//   feat_col/feat_code (one-hot membership test or numeric passthrough).
// DIRECT=true: isolation forest — node.feat indexes the numeric columns.
template <bool DIRECT>
__global__ __launch_bounds__(BLOCK) void forest_kernel(
    const short* __restrict__ codes,    // [B, N_CAT]
    const float* __restrict__ nums,     // [B, N_NUM]
    const float* __restrict__ medians,  // [N_NUM]
    const int4* __restrict__ nodes,     // [n_nodes] {feat, bits, left, right}
    const int* __restrict__ tree_off,   // [T+1]
    int n_trees,
    const int* __restrict__ feat_col,   // [F] (unused when DIRECT)
    const int* __restrict__ feat_code,  // [F] (unused when DIRECT)
    int n_rows,
    double* __restrict__ acc)           // [B], pre-zeroed
{
  // Column-major LDS staging: lane-consecutive addresses per column access
  // are stride-1 → conflict-free; dynamic per-node column indexing stays in
  // LDS instead of spilling a register-indexed array to scratch.
  __shared__ short s_codes[N_CAT * BLOCK];
  __shared__ float s_nums[N_NUM * BLOCK];

  const int tid = threadIdx.x;
  const int row = blockIdx.x * BLOCK + tid;
  if (row < n_rows) {
    if (!DIRECT) {
#pragma unroll
      for (int c = 0; c < N_CAT; ++c) s_codes[c * BLOCK + tid] = codes[row * N_CAT + c];
    }
#pragma unroll
    for (int c = 0; c < N_NUM; ++c) {
      const float v = nums[row * N_NUM + c];
      s_nums[c * BLOCK + tid] = isnan(v) ? medians[c] : v;  // fused imputation
    }
  }
  // No __syncthreads(): each thread only reads its own LDS slots.
  if (row >= n_rows) return;

  double local = 0.0;
  for (int t = blockIdx.y; t < n_trees; t += gridDim.y) {
    const int base = tree_off[t];
    int4 nd = nodes[base];
    while (nd.x >= 0) {
      float v;
      if (DIRECT) {
        v = s_nums[nd.x * BLOCK + tid];
      } else {
        const int col = feat_col[nd.x];
        const int code = feat_code[nd.x];
        v = (code >= 0) ? ((s_codes[col * BLOCK + tid] == (short)code) ? 1.0f : 0.0f)
                        : s_nums[col * BLOCK + tid];
      }
      const float thr = __int_as_float(nd.y);
      nd = nodes[base + ((v <= thr) ? nd.z : nd.w)];
    }
    local += (double)__int_as_float(nd.y);  // leaf payload
  }
  atomicAdd(&acc[row], local);
}

// ILP variant: each thread walks TWO trees concurrently — two independent
// node-gather chains per lane double the memory-level parallelism of the
// latency-bound traversal (A/B'd against forest_kernel via kernel_micro).
template <bool DIRECT>
__global__ __launch_bounds__(BLOCK) void forest_kernel_ilp(
    const short* __restrict__ codes,
    const float* __restrict__ nums,
    const float* __restrict__ medians,
    const int4* __restrict__ nodes,
    const int* __restrict__ tree_off,
    int n_trees,
    const int* __restrict__ feat_col,
    const int* __restrict__ feat_code,
    int n_rows,
    double* __restrict__ acc)
{
  __shared__ short s_codes[N_CAT * BLOCK];
  __shared__ float s_nums[N_NUM * BLOCK];
  const int tid = threadIdx.x;
  const int row = blockIdx.x * BLOCK + tid;
  if (row < n_rows) {
    if (!DIRECT) {
#pragma unroll
      for (int c = 0; c < N_CAT; ++c) s_codes[c * BLOCK + tid] = codes[row * N_CAT + c];
    }
#pragma unroll
    for (int c = 0; c < N_NUM; ++c) {
      const float v = nums[row * N_NUM + c];
      s_nums[c * BLOCK + tid] = isnan(v) ? medians[c] : v;
    }
  }
  if (row >= n_rows) return;

  auto value_of = [&](int f) -> float {
    if (DIRECT) return s_nums[f * BLOCK + tid];
    const int col = feat_col[f];
    const int code = feat_code[f];
    return (code >= 0) ? ((s_codes[col * BLOCK + tid] == (short)code) ? 1.0f : 0.0f)
                       : s_nums[col * BLOCK + tid];
  };

  double local = 0.0;
  for (int t = blockIdx.y; t < n_trees; t += 2 * gridDim.y) {
    const int ta = t;
    const int tb = t + gridDim.y;
    const int basea = tree_off[ta];
    int4 na = nodes[basea];
    bool la = true;
    int baseb = 0;
    int4 nb;
    bool lb = tb < n_trees;
    if (lb) { baseb = tree_off[tb]; nb = nodes[baseb]; }
    while (la || lb) {
      if (la) {
        if (na.x >= 0) {
          const float v = value_of(na.x);
          na = nodes[basea + ((v <= __int_as_float(na.y)) ? na.z : na.w)];
        } else {
          local += (double)__int_as_float(na.y);
          la = false;
        }
      }
      if (lb) {
        if (nb.x >= 0) {
          const float v = value_of(nb.x);
          nb = nodes[baseb + ((v <= __int_as_float(nb.y)) ? nb.z : nb.w)];
        } else {
          local += (double)__int_as_float(nb.y);
          lb = false;
        }
      }
    }
  }
  atomicAdd(&acc[row], local);
}

// Dual-forest fusion: classifier (one-hot resolution) and isolation forest
// (direct numeric features) in ONE launch — grid-y is split into
// chunks_cls classifier chunks followed by iforest chunks, each block
// running the same 2-tree-ILP walk against its forest's tables. The two
// forests were two serial launches in round 1 (~9.4 µs of iforest after
// ~33 µs of classifier at b=1024); one grid lets the scheduler run both
// concurrently with NO new graph edges (a 3-stream fork was measured
// SLOWER — the extra cross-stream edges cost ~20 µs/replay, see
// kernel_tuning.md).
template <bool DIRECT>
__device__ __forceinline__ double forest_walk_ilp2(
    const short* __restrict__ s_codes,
    const float* __restrict__ s_nums,
    int tid,
    const int4* __restrict__ nodes,
    const int* __restrict__ tree_off,
    int n_trees,
    const int* __restrict__ feat_col,
    const int* __restrict__ feat_code,
    int chunk,
    int n_chunks)
{
  auto value_of = [&](int f) -> float {
    if (DIRECT) return s_nums[f * BLOCK + tid];
    const int col = feat_col[f];
    const int code = feat_code[f];
    return (code >= 0) ? ((s_codes[col * BLOCK + tid] == (short)code) ? 1.0f : 0.0f)
                       : s_nums[col * BLOCK + tid];
  };
  double local = 0.0;
  for (int t = chunk; t < n_trees; t += 2 * n_chunks) {
    const int ta = t;
    const int tb = t + n_chunks;
    const int basea = tree_off[ta];
    int4 na = nodes[basea];
    bool la = true;
    int baseb = 0;
    int4 nb;
    bool lb = tb < n_trees;
    if (lb) { baseb = tree_off[tb]; nb = nodes[baseb]; }
    while (la || lb) {
      if (la) {
        if (na.x >= 0) {
          const float v = value_of(na.x);
          na = nodes[basea + ((v <= __int_as_float(na.y)) ? na.z : na.w)];
        } else {
          local += (double)__int_as_float(na.y);
          la = false;
        }
      }
      if (lb) {
        if (nb.x >= 0) {
          const float v = value_of(nb.x);
          nb = nodes[baseb + ((v <= __int_as_float(nb.y)) ? nb.z : nb.w)];
        } else {
          local += (double)__int_as_float(nb.y);
          lb = false;
        }
      }
    }
  }
  return local;
}

__global__ __launch_bounds__(BLOCK) void forest_kernel_dual(
    const short* __restrict__ codes,
    const float* __restrict__ nums,
    const float* __restrict__ medians,
    const int4* __restrict__ cls_nodes,
    const int* __restrict__ cls_off,
    int t_cls,
    const int* __restrict__ feat_col,
    const int* __restrict__ feat_code,
    const int4* __restrict__ if_nodes,
    const int* __restrict__ if_off,
    int t_if,
    int chunks_cls,
    int n_rows,
    double* __restrict__ acc_cls,
    double* __restrict__ acc_if)
{
  __shared__ short s_codes[N_CAT * BLOCK];
  __shared__ float s_nums[N_NUM * BLOCK];
  const int tid = threadIdx.x;
  const int row = blockIdx.x * BLOCK + tid;
  const bool is_cls = (int)blockIdx.y < chunks_cls;
  if (row < n_rows) {
    if (is_cls) {
#pragma unroll
      for (int c = 0; c < N_CAT; ++c) s_codes[c * BLOCK + tid] = codes[row * N_CAT + c];
    }
#pragma unroll
    for (int c = 0; c < N_NUM; ++c) {
      const float v = nums[row * N_NUM + c];
      s_nums[c * BLOCK + tid] = isnan(v) ? medians[c] : v;
    }
  }
  if (row >= n_rows) return;
  if (is_cls) {
    const double local = forest_walk_ilp2<false>(
        s_codes, s_nums, tid, cls_nodes, cls_off, t_cls, feat_col, feat_code,
        blockIdx.y, chunks_cls);
    atomicAdd(&acc_cls[row], local);
  } else {
    const double local = forest_walk_ilp2<true>(
        s_codes, s_nums, tid, if_nodes, if_off, t_if, nullptr, nullptr,
        blockIdx.y - chunks_cls, gridDim.y - chunks_cls);
    atomicAdd(&acc_if[row], local);
  }
}

// 4-tree ILP + optional transposed grid (blockIdx.x = tree chunk so the
// dispatcher's id%8 XCD placement gives adjacent chunks to different XCDs,
// keeping each XCD's L2 on a tree subset). Experimental A/B variants.
template <bool DIRECT, int ILP, bool SWAP_GRID>
__global__ __launch_bounds__(BLOCK) void forest_kernel_ilpN(
    const short* __restrict__ codes,
    const float* __restrict__ nums,
    const float* __restrict__ medians,
    const int4* __restrict__ nodes,
    const int* __restrict__ tree_off,
    int n_trees,
    const int* __restrict__ feat_col,
    const int* __restrict__ feat_code,
    int n_rows,
    double* __restrict__ acc)
{
  __shared__ short s_codes[N_CAT * BLOCK];
  __shared__ float s_nums[N_NUM * BLOCK];
  const int tid = threadIdx.x;
  const int row_blk = SWAP_GRID ? blockIdx.y : blockIdx.x;
  const int chunk = SWAP_GRID ? blockIdx.x : blockIdx.y;
  const int n_chunks = SWAP_GRID ? gridDim.x : gridDim.y;
  const int row = row_blk * BLOCK + tid;
  if (row < n_rows) {
    if (!DIRECT) {
#pragma unroll
      for (int c = 0; c < N_CAT; ++c) s_codes[c * BLOCK + tid] = codes[row * N_CAT + c];
    }
#pragma unroll
    for (int c = 0; c < N_NUM; ++c) {
      const float v = nums[row * N_NUM + c];
      s_nums[c * BLOCK + tid] = isnan(v) ? medians[c] : v;
    }
  }
  if (row >= n_rows) return;

  auto value_of = [&](int f) -> float {
    if (DIRECT) return s_nums[f * BLOCK + tid];
    const int col = feat_col[f];
    const int code = feat_code[f];
    return (code >= 0) ? ((s_codes[col * BLOCK + tid] == (short)code) ? 1.0f : 0.0f)
                       : s_nums[col * BLOCK + tid];
  };

  double local = 0.0;
  for (int t = chunk; t < n_trees; t += ILP * n_chunks) {
    int4 nd[ILP];
    int base[ILP];
    bool live[ILP];
#pragma unroll
    for (int k = 0; k < ILP; ++k) {
      const int tk = t + k * n_chunks;
      live[k] = tk < n_trees;
      if (live[k]) {
        base[k] = tree_off[tk];
        nd[k] = nodes[base[k]];
      }
    }
    bool any = true;
    while (any) {
      any = false;
#pragma unroll
      for (int k = 0; k < ILP; ++k) {
        if (!live[k]) continue;
        if (nd[k].x >= 0) {
          const float v = value_of(nd[k].x);
          nd[k] = nodes[base[k] + ((v <= __int_as_float(nd[k].y)) ? nd[k].z : nd[k].w)];
          any = true;
        } else {
          local += (double)__int_as_float(nd[k].y);
          live[k] = false;
        }
      }
    }
  }
  atomicAdd(&acc[row], local);
}

__global__ __launch_bounds__(BLOCK) void finalize_kernel(
    double* __restrict__ cls_acc,
    double* __restrict__ if_acc,
    int n_rows,
    int cls_kind,        // 0 = RF leaf-fraction mean; 1 = GBT logit sum
    double inv_n_trees,
    double cls_bias,
    double if_denom,
    double if_offset,
    double if_threshold,
    double* __restrict__ proba,
    double* __restrict__ iscore,
    double* __restrict__ outlier)
{
  const int i = blockIdx.x * BLOCK + threadIdx.x;
  if (i >= n_rows) return;
  proba[i] = cls_kind ? 1.0 / (1.0 + exp(-(cls_acc[i] + cls_bias)))
                      : cls_acc[i] * inv_n_trees;
  // alibi instance_score = -(sklearn decision_function)
  //                      = 2^(-mean_depth_sum/denom) + offset_
  const double anomaly = exp2(-if_acc[i] / if_denom);
  const double s = anomaly + if_offset;
  iscore[i] = s;
  outlier[i] = (s > if_threshold) ? 1.0 : 0.0;
  // zero-after-read: the accumulators start all-zero (torch::zeros at
  // session init) and every finalize re-zeros the rows it consumed, so no
  // per-request memset node is needed in the graph regardless of how batch
  // sizes interleave across captured shapes
  cls_acc[i] = 0.0;
  if_acc[i] = 0.0;
}

// ---------------------------------------------------------------------------
// Drift statistics
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(BLOCK) void cat_hist_kernel(
    const short* __restrict__ codes,  // [B, N_CAT]
    int n_rows,
    const int* __restrict__ cat_off,  // [N_CAT+1]
    int total_bins,
    int* __restrict__ hist)           // [total_bins], pre-zeroed
{
  extern __shared__ int s_hist[];
  for (int i = threadIdx.x; i < total_bins; i += blockDim.x) s_hist[i] = 0;
  __syncthreads();
  __shared__ int s_off[N_CAT + 1];
  if (threadIdx.x <= N_CAT) s_off[threadIdx.x] = cat_off[threadIdx.x];
  __syncthreads();

  for (int r = blockIdx.x * blockDim.x + threadIdx.x; r < n_rows;
       r += gridDim.x * blockDim.x) {
#pragma unroll
    for (int c = 0; c < N_CAT; ++c) {
      const int lo = s_off[c];
      const int nbins = s_off[c + 1] - lo;
      const int code = codes[r * N_CAT + c];
      // unknown/unseen category (-1) lands in the trailing "unseen" bin,
      // matching creditcore.ops.cpu_ref.drift_stats_cpu
      const int bin = (code < 0) ? (nbins - 1) : code;
      atomicAdd(&s_hist[lo + bin], 1);
    }
  }
  __syncthreads();
  if (gridDim.x == 1) {
    // single-block launch (small batches): plain stores — the output needs
    // no pre-zero memset node in the graph
    for (int i = threadIdx.x; i < total_bins; i += blockDim.x)
      hist[i] = s_hist[i];
  } else {
    for (int i = threadIdx.x; i < total_bins; i += blockDim.x)
      if (s_hist[i] != 0) atomicAdd(&hist[i], s_hist[i]);
  }
}

// One block per numeric feature. Sorts the (imputed) batch column in LDS with
// a bitonic sort, then evaluates sup|F_ref - F_batch| at every batch point's
// left/right limits (the extrema of two step CDFs — same algorithm as
// creditcore.models.drift.ks_2samp_d). CDF arithmetic in f64 to match the
// CPU reference bitwise-closely. When the reference column fits the
// remaining LDS (160 KiB/CU on gfx950) it is staged there too, so the
// per-element binary searches hit LDS instead of bouncing off L2.
template <int BS>
__global__ __launch_bounds__(BS) void ks_kernel_t(
    const float* __restrict__ nums,       // [B, n_cols]
    const float* __restrict__ medians,    // [n_cols]
    int n_cols,                           // feature count == gridDim.x
    int n_rows,
    int m_pow2,                           // next pow2 >= n_rows
    int ref_lds,                          // 1 => stage ref column in LDS
    const float* __restrict__ ref_sorted, // concatenated per-feature refs
    const int64_t* __restrict__ rs_off,   // [n_cols+1] (i64: offsets exceed
                                          // 2^31 for HBM-scale references)
    float* __restrict__ ks_d)             // [n_cols]
{
  extern __shared__ float s_vals[];  // [m_pow2] batch | [n_ref] staged ref
  const int j = blockIdx.x;
  const int m = n_rows;

  for (int i = threadIdx.x; i < m_pow2; i += blockDim.x) {
    float v = INFINITY;
    if (i < m) {
      v = nums[i * n_cols + j];
      if (isnan(v)) v = medians[j];
    }
    s_vals[i] = v;
  }
  const int64_t ref_lo = rs_off[j];
  const int n_ref_j = (int)(rs_off[j + 1] - ref_lo);
  if (ref_lds) {
    float* s_ref = s_vals + m_pow2;
    for (int i = threadIdx.x; i < n_ref_j; i += blockDim.x)
      s_ref[i] = ref_sorted[ref_lo + i];
  }
  __syncthreads();

  // bitonic sort (ascending)
  for (int k = 2; k <= m_pow2; k <<= 1) {
    for (int s = k >> 1; s > 0; s >>= 1) {
      for (int i = threadIdx.x; i < m_pow2; i += blockDim.x) {
        const int p = i ^ s;
        if (p > i) {
          const float a = s_vals[i];
          const float b = s_vals[p];
          const bool up = ((i & k) == 0);
          if ((a > b) == up) {
            s_vals[i] = b;
            s_vals[p] = a;
          }
        }
      }
      __syncthreads();
    }
  }

  const int n = n_ref_j;
  const float* __restrict__ ref = ref_lds ? (s_vals + m_pow2) : (ref_sorted + ref_lo);

  double dmax = 0.0;
  for (int i = threadIdx.x; i < m; i += blockDim.x) {
    const float b = s_vals[i];
    int l = 0, r = n;  // lower_bound in ref
    while (l < r) {
      const int mid = (l + r) >> 1;
      if (ref[mid] < b) l = mid + 1; else r = mid;
    }
    const int sl = l;
    r = n;  // upper_bound in ref (resume from sl)
    while (l < r) {
      const int mid = (l + r) >> 1;
      if (ref[mid] <= b) l = mid + 1; else r = mid;
    }
    const int sr = l;
    // F_batch one-sided limits at b: the tie run's bounds in the sorted
    // batch, not the per-element rank (matches models/drift.ks_2samp_d).
    int bl = 0;
    r = i;  // lower_bound of b within s_vals[0..i]
    while (bl < r) {
      const int mid = (bl + r) >> 1;
      if (s_vals[mid] < b) bl = mid + 1; else r = mid;
    }
    int br = i + 1;
    r = m;  // upper_bound of b within s_vals[i+1..m)
    while (br < r) {
      const int mid = (br + r) >> 1;
      if (s_vals[mid] <= b) br = mid + 1; else r = mid;
    }
    const double fl = fabs((double)sl / n - (double)bl / m);
    const double fr = fabs((double)sr / n - (double)br / m);
    dmax = fmax(dmax, fmax(fl, fr));
  }

  // wave reduce (64-wide) then cross-wave via LDS (reuse s_vals after sync)
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    dmax = fmax(dmax, __shfl_down(dmax, off, 64));
  __syncthreads();
  float* s_red = s_vals;  // one slot per wave
  const int wave = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) s_red[wave] = (float)dmax;
  __syncthreads();
  if (threadIdx.x == 0) {
    float d = s_red[0];
    for (int w = 1; w < (int)(blockDim.x >> 6); ++w) d = fmaxf(d, s_red[w]);
    ks_d[j] = d;
  }
}

// Small-batch K-S variant: instead of a bitonic sort (log^2(B)/2 barriered
// LDS passes — 55 stages at B=1024, ~39 us with only n_cols workgroups on
// the chip), count each element's strict / non-strict rank directly with an
// O(B^2) LDS sweep. The inner loop reads s_vals[k] at the same k across the
// whole wavefront (LDS broadcast, no bank conflicts, no barriers), so at
// B<=2048 the quadratic sweep is several times faster than the sort's
// barrier chain. Output is bitwise-identical to ks_kernel_t: lt/le ARE the
// sorted path's tie-run lower/upper bounds, and the ref binary searches and
// f64 CDF arithmetic are the same (models/drift.ks_2samp_d semantics).
template <int BS>
__global__ __launch_bounds__(BS) void ks_count_kernel_t(
    const float* __restrict__ nums,       // [B, n_cols]
    const float* __restrict__ medians,    // [n_cols]
    int n_cols, int n_rows,
    const float* __restrict__ ref_sorted, const int64_t* __restrict__ rs_off,
    float* __restrict__ ks_d)
{
  extern __shared__ float s_vals[];  // [n_rows] imputed batch column
  const int j = blockIdx.x;
  const int m = n_rows;
  for (int i = threadIdx.x; i < m; i += blockDim.x) {
    float v = nums[i * n_cols + j];
    if (isnan(v)) v = medians[j];
    s_vals[i] = v;
  }
  const int64_t ref_lo = rs_off[j];
  const int n = (int)(rs_off[j + 1] - ref_lo);
  const float* __restrict__ ref = ref_sorted + ref_lo;
  __syncthreads();

  double dmax = 0.0;
  for (int i = threadIdx.x; i < m; i += blockDim.x) {
    const float x = s_vals[i];
    int lt = 0, le = 0;  // # batch elements < x / <= x
    int k = 0;
    for (; k + 3 < m; k += 4) {
      const float v0 = s_vals[k], v1 = s_vals[k + 1];
      const float v2 = s_vals[k + 2], v3 = s_vals[k + 3];
      lt += (int)(v0 < x) + (int)(v1 < x) + (int)(v2 < x) + (int)(v3 < x);
      le += (int)(v0 <= x) + (int)(v1 <= x) + (int)(v2 <= x) + (int)(v3 <= x);
    }
    for (; k < m; ++k) {
      const float v = s_vals[k];
      lt += (int)(v < x);
      le += (int)(v <= x);
    }
    int l = 0, r = n;  // lower_bound in ref
    while (l < r) {
      const int mid = (l + r) >> 1;
      if (ref[mid] < x) l = mid + 1; else r = mid;
    }
    const int sl = l;
    r = n;  // upper_bound in ref (resume from sl)
    while (l < r) {
      const int mid = (l + r) >> 1;
      if (ref[mid] <= x) l = mid + 1; else r = mid;
    }
    const int sr = l;
    const double fl = fabs((double)sl / n - (double)lt / m);
    const double fr = fabs((double)sr / n - (double)le / m);
    dmax = fmax(dmax, fmax(fl, fr));
  }

#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    dmax = fmax(dmax, __shfl_down(dmax, off, 64));
  __syncthreads();
  float* s_red = s_vals;  // one slot per wave (launch smem floors at BS/64)
  const int wave = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) s_red[wave] = (float)dmax;
  __syncthreads();
  if (threadIdx.x == 0) {
    float d = s_red[0];
    for (int w = 1; w < (int)(blockDim.x >> 6); ++w) d = fmaxf(d, s_red[w]);
    ks_d[j] = d;
  }
}

// counting beats the sort only while the quadratic sweep stays tiny:
// measured on MI355X, counting is 51 us at B=1024 vs the bitonic's 38.6 us
// (the LDS-broadcast sweep is ALU-bound, not barrier-bound), but wins
// clearly on the small batches the single-call latency path serves
constexpr int KS_COUNT_MAX_ROWS = 256;

// Multi-block counting K-S: the single-block kernels occupy only n_cols
// (=14) of the 256 CUs. This variant splits each column's element range
// over gridDim.y blocks — every block stages the full imputed column in
// LDS (B ≤ 16384 ⇒ ≤ 64 KiB) and sweeps its slice with the same
// LDS-broadcast quadratic count, so ~200+ workgroups fill the chip and
// the O(B²) work parallelizes. Per-(column, block) partial maxima land in
// `scratch`; ks_count_reduce_kernel folds them into ks_d. lt/le counting
// is identical to the single-block kernels (sorted-path tie-run bounds),
// and f64→f32 casting commutes with max, so the result is bitwise equal
// to ks_kernel_t's.
template <int BS>
__global__ __launch_bounds__(BS) void ks_count_mb_kernel_t(
    const float* __restrict__ nums,       // [B, n_cols]
    const float* __restrict__ medians,    // [n_cols]
    int n_cols, int n_rows,
    const float* __restrict__ ref_sorted, const int64_t* __restrict__ rs_off,
    float* __restrict__ scratch)          // [n_cols, gridDim.y]
{
  extern __shared__ float s_vals[];  // [n_rows] imputed batch column
  const int j = blockIdx.x;
  const int m = n_rows;
  for (int i = threadIdx.x; i < m; i += blockDim.x) {
    float v = nums[i * n_cols + j];
    if (isnan(v)) v = medians[j];
    s_vals[i] = v;
  }
  const int64_t ref_lo = rs_off[j];
  const int n = (int)(rs_off[j + 1] - ref_lo);
  const float* __restrict__ ref = ref_sorted + ref_lo;
  __syncthreads();

  // this block's element slice: [e0, e1)
  const int per = (m + gridDim.y - 1) / gridDim.y;
  const int e0 = blockIdx.y * per;
  const int e1 = min(m, e0 + per);

  double dmax = 0.0;
  for (int i = e0 + threadIdx.x; i < e1; i += blockDim.x) {
    const float x = s_vals[i];
    int lt = 0, le = 0;
    int k = 0;
    for (; k + 3 < m; k += 4) {
      const float v0 = s_vals[k], v1 = s_vals[k + 1];
      const float v2 = s_vals[k + 2], v3 = s_vals[k + 3];
      lt += (int)(v0 < x) + (int)(v1 < x) + (int)(v2 < x) + (int)(v3 < x);
      le += (int)(v0 <= x) + (int)(v1 <= x) + (int)(v2 <= x) + (int)(v3 <= x);
    }
    for (; k < m; ++k) {
      const float v = s_vals[k];
      lt += (int)(v < x);
      le += (int)(v <= x);
    }
    int l = 0, r = n;
    while (l < r) {
      const int mid = (l + r) >> 1;
      if (ref[mid] < x) l = mid + 1; else r = mid;
    }
    const int sl = l;
    r = n;
    while (l < r) {
      const int mid = (l + r) >> 1;
      if (ref[mid] <= x) l = mid + 1; else r = mid;
    }
    const int sr = l;
    const double fl = fabs((double)sl / n - (double)lt / m);
    const double fr = fabs((double)sr / n - (double)le / m);
    dmax = fmax(dmax, fmax(fl, fr));
  }

#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    dmax = fmax(dmax, __shfl_down(dmax, off, 64));
  __syncthreads();
  float* s_red = s_vals;  // one slot per wave (smem floors at BS/64)
  const int wave = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) s_red[wave] = (float)dmax;
  __syncthreads();
  if (threadIdx.x == 0) {
    float d = s_red[0];
    for (int w = 1; w < (int)(blockDim.x >> 6); ++w) d = fmaxf(d, s_red[w]);
    scratch[(int64_t)j * gridDim.y + blockIdx.y] = d;
  }
}

__global__ __launch_bounds__(64) void ks_count_reduce_kernel(
    const float* __restrict__ scratch,  // [n_cols, g]
    int g, int n_cols,
    float* __restrict__ ks_d)
{
  const int j = blockIdx.x * 64 + threadIdx.x;
  if (j >= n_cols) return;
  float d = 0.0f;
  for (int k = 0; k < g; ++k) d = fmaxf(d, scratch[(int64_t)j * g + k]);
  ks_d[j] = d;
}

// blocks per column for the multi-block counting path: ~128 elements per
// block keeps every block's sweep a few microseconds while n_cols*g
// workgroups fill the 256 CUs
inline int ks_mb_blocks(int b) {
  int g = (b + 127) / 128;
  if (g < 1) g = 1;
  if (g > 128) g = 128;
  return g;
}
constexpr int KS_MB_MAX_G = 128;

// ---------------------------------------------------------------------------
// Host launchers
// ---------------------------------------------------------------------------

namespace {

inline int ceil_div(int a, int b) { return (a + b - 1) / b; }

void check_inputs(const torch::Tensor& codes, const torch::Tensor& nums) {
  TORCH_CHECK(codes.is_cuda() && nums.is_cuda(), "inputs must be on GPU");
  TORCH_CHECK(codes.scalar_type() == torch::kInt16, "codes must be int16");
  TORCH_CHECK(nums.scalar_type() == torch::kFloat32, "nums must be float32");
  TORCH_CHECK(codes.is_contiguous() && nums.is_contiguous(), "inputs must be contiguous");
  TORCH_CHECK(codes.size(1) == N_CAT && nums.size(1) == N_NUM, "bad feature counts");
  TORCH_CHECK(codes.size(0) == nums.size(0), "row count mismatch");
}

}  // namespace

std::vector<torch::Tensor> score_forest_pipeline(
    torch::Tensor codes, torch::Tensor nums,
    torch::Tensor cls_nodes, torch::Tensor cls_off,
    torch::Tensor feat_col, torch::Tensor feat_code,
    torch::Tensor medians, int64_t n_onehot,
    torch::Tensor if_nodes, torch::Tensor if_off,
    double if_denom, double if_offset, double if_threshold)
{
  (void)n_onehot;  // encoded in feat_code (>=0 ⇔ one-hot)
  check_inputs(codes, nums);
  const int B = (int)codes.size(0);
  const int t_cls = (int)cls_off.size(0) - 1;
  const int t_if = (int)if_off.size(0) - 1;

  auto opts_f64 = torch::TensorOptions().dtype(torch::kFloat64).device(codes.device());
  auto cls_acc = torch::zeros({B}, opts_f64);
  auto if_acc = torch::zeros({B}, opts_f64);
  auto proba = torch::empty({B}, opts_f64);
  auto iscore = torch::empty({B}, opts_f64);
  auto outlier = torch::empty({B}, opts_f64);

  hipStream_t stream = c10::hip::getCurrentHIPStream();
  const int row_blocks = ceil_div(B, BLOCK);
  // enough blocks to cover the 256 CUs / 8 XCDs even for small batches
  auto tree_chunks = [&](int t) {
    int c = ceil_div(2048, row_blocks);
    return std::max(1, std::min(c, t));
  };

  dim3 g_cls(row_blocks, tree_chunks(t_cls));
  hipLaunchKernelGGL((forest_kernel<false>), g_cls, dim3(BLOCK), 0, stream,
      codes.data_ptr<short>(), nums.data_ptr<float>(), medians.data_ptr<float>(),
      reinterpret_cast<const int4*>(cls_nodes.data_ptr<int>()),
      cls_off.data_ptr<int>(), t_cls,
      feat_col.data_ptr<int>(), feat_code.data_ptr<int>(),
      B, cls_acc.data_ptr<double>());

  dim3 g_if(row_blocks, tree_chunks(t_if));
  hipLaunchKernelGGL((forest_kernel<true>), g_if, dim3(BLOCK), 0, stream,
      codes.data_ptr<short>(), nums.data_ptr<float>(), medians.data_ptr<float>(),
      reinterpret_cast<const int4*>(if_nodes.data_ptr<int>()),
      if_off.data_ptr<int>(), t_if,
      nullptr, nullptr,
      B, if_acc.data_ptr<double>());

  hipLaunchKernelGGL(finalize_kernel, dim3(row_blocks), dim3(BLOCK), 0, stream,
      cls_acc.data_ptr<double>(), if_acc.data_ptr<double>(), B,
      /*cls_kind=*/0, 1.0 / (double)t_cls, 0.0, if_denom, if_offset, if_threshold,
      proba.data_ptr<double>(), iscore.data_ptr<double>(), outlier.data_ptr<double>());
  HIP_CHECK(hipGetLastError());

  return {proba, iscore, outlier};
}

torch::Tensor forest_ilp_bench(
    torch::Tensor codes, torch::Tensor nums,
    torch::Tensor nodes, torch::Tensor off,
    torch::Tensor feat_col, torch::Tensor feat_code,
    torch::Tensor medians, int64_t use_ilp)
{
  check_inputs(codes, nums);
  const int B = (int)codes.size(0);
  const int T = (int)off.size(0) - 1;
  auto acc = torch::zeros({B},
      torch::TensorOptions().dtype(torch::kFloat64).device(codes.device()));
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  const int row_blocks = ceil_div(B, BLOCK);
  int chunks = std::max(1, std::min(ceil_div(2048, row_blocks), T));
  if (use_ilp) {
    const int ilp = (use_ilp == 2 || use_ilp == 4) ? 4 : 2;
    const bool swap = use_ilp >= 3;
    chunks = std::max(1, std::min(ceil_div(2048, row_blocks), (T + ilp - 1) / ilp));
    dim3 grid = swap ? dim3(chunks, row_blocks) : dim3(row_blocks, chunks);
    auto args = std::make_tuple(
        codes.data_ptr<short>(), nums.data_ptr<float>(), medians.data_ptr<float>(),
        reinterpret_cast<const int4*>(nodes.data_ptr<int>()), off.data_ptr<int>(), T,
        feat_col.data_ptr<int>(), feat_code.data_ptr<int>(), B, acc.data_ptr<double>());
    auto launch = [&](auto kernel) {
      std::apply([&](auto... a) {
        hipLaunchKernelGGL(kernel, grid, dim3(BLOCK), 0, stream, a...);
      }, args);
    };
    if (use_ilp == 1) launch(forest_kernel_ilpN<false, 2, false>);
    else if (use_ilp == 2) launch(forest_kernel_ilpN<false, 4, false>);
    else if (use_ilp == 3) launch(forest_kernel_ilpN<false, 2, true>);
    else launch(forest_kernel_ilpN<false, 4, true>);
  } else {
    hipLaunchKernelGGL((forest_kernel<false>), dim3(row_blocks, chunks),
        dim3(BLOCK), 0, stream,
        codes.data_ptr<short>(), nums.data_ptr<float>(), medians.data_ptr<float>(),
        reinterpret_cast<const int4*>(nodes.data_ptr<int>()), off.data_ptr<int>(), T,
        feat_col.data_ptr<int>(), feat_code.data_ptr<int>(), B, acc.data_ptr<double>());
  }
  HIP_CHECK(hipGetLastError());
  return acc;
}

std::vector<torch::Tensor> drift_stats(
    torch::Tensor codes, torch::Tensor nums, torch::Tensor medians,
    torch::Tensor ref_sorted, torch::Tensor rs_off, torch::Tensor cat_off,
    int64_t total_bins)
{
  check_inputs(codes, nums);
  const int B = (int)codes.size(0);
  TORCH_CHECK(B <= MAX_DRIFT_ROWS,
      "drift batch too large for the K-S LDS sort (cap it host-side): ", B);

  auto hist = torch::zeros({total_bins},
      torch::TensorOptions().dtype(torch::kInt32).device(codes.device()));
  auto ks_d = torch::empty({N_NUM},
      torch::TensorOptions().dtype(torch::kFloat32).device(codes.device()));
  auto rs_off64 = rs_off.to(torch::kInt64);

  hipStream_t stream = c10::hip::getCurrentHIPStream();

  const int hist_blocks = std::min(ceil_div(B, BLOCK), 1024);
  hipLaunchKernelGGL(cat_hist_kernel, dim3(hist_blocks), dim3(BLOCK),
      (size_t)total_bins * sizeof(int), stream,
      codes.data_ptr<short>(), B, cat_off.data_ptr<int>(), (int)total_bins,
      hist.data_ptr<int>());

  if (B <= KS_COUNT_MAX_ROWS) {
    const size_t smem = (size_t)std::max(B, BLOCK / 64) * sizeof(float);
    hipLaunchKernelGGL((ks_count_kernel_t<256>), dim3(N_NUM), dim3(BLOCK),
        smem, stream,
        nums.data_ptr<float>(), medians.data_ptr<float>(), N_NUM, B,
        ref_sorted.data_ptr<float>(), rs_off64.data_ptr<int64_t>(),
        ks_d.data_ptr<float>());
  } else {
    int m_pow2 = 1;
    while (m_pow2 < B) m_pow2 <<= 1;
    // LDS must also hold one cross-wave reduction slot per wave
    m_pow2 = std::max(m_pow2, BLOCK / 64);
    hipLaunchKernelGGL((ks_kernel_t<256>), dim3(N_NUM), dim3(BLOCK),
        (size_t)m_pow2 * sizeof(float), stream,
        nums.data_ptr<float>(), medians.data_ptr<float>(), N_NUM, B, m_pow2,
        /*ref_lds=*/0, ref_sorted.data_ptr<float>(),
        rs_off64.data_ptr<int64_t>(), ks_d.data_ptr<float>());
  }
  HIP_CHECK(hipGetLastError());

  return {hist, ks_d};
}

// ---------------------------------------------------------------------------
// Dense wide-tabular family (BASELINE config 5: 10M x 1k synthetic).
// dense_score_kernel: fused impute + logistic linear score + robust-z
// outlier flag, one wavefront per row (64 lanes stride the F features with
// coalesced loads; wave shuffle-reduce for the dot product and max-z).
// Drift for this family reuses ks_kernel with n_cols = F and the per-feature
// sorted reference resident in HBM (the 288 GB sizing: a 10M x 1k f32
// reference is 40 GB per GPU).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(BLOCK) void dense_score_kernel(
    const float* __restrict__ x,        // [B, F]
    const float* __restrict__ medians,  // [F]
    const float* __restrict__ inv_scale,// [F] 1/(IQR) robust scale
    const float* __restrict__ w,        // [F]
    float bias,
    float z_threshold,
    int n_rows,
    int n_feat,
    double* __restrict__ proba,         // [B]
    double* __restrict__ iscore,        // [B] max robust z
    double* __restrict__ outlier)       // [B]
{
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int row = blockIdx.x * (BLOCK / 64) + wave;
  if (row >= n_rows) return;
  const float* xr = x + (size_t)row * n_feat;

  float acc = 0.0f;
  float zmax = 0.0f;
  for (int f = lane; f < n_feat; f += 64) {
    float v = xr[f];
    const float med = medians[f];
    if (isnan(v)) v = med;  // fused median imputation
    acc += v * w[f];
    const float z = fabsf(v - med) * inv_scale[f];
    zmax = fmaxf(zmax, z);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    acc += __shfl_down(acc, off, 64);
    zmax = fmaxf(zmax, __shfl_down(zmax, off, 64));
  }
  if (lane == 0) {
    const double logit = (double)acc + (double)bias;
    proba[row] = 1.0 / (1.0 + exp(-logit));
    iscore[row] = (double)zmax;
    outlier[row] = (zmax > z_threshold) ? 1.0 : 0.0;
  }
}

std::vector<torch::Tensor> dense_score(
    torch::Tensor x, torch::Tensor medians, torch::Tensor inv_scale,
    torch::Tensor w, double bias, double z_threshold)
{
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kFloat32 && x.is_contiguous());
  const int B = (int)x.size(0);
  const int F = (int)x.size(1);
  TORCH_CHECK(medians.numel() == F && w.numel() == F && inv_scale.numel() == F);
  auto opts = torch::TensorOptions().dtype(torch::kFloat64).device(x.device());
  auto proba = torch::empty({B}, opts);
  auto iscore = torch::empty({B}, opts);
  auto outlier = torch::empty({B}, opts);
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  const int rows_per_block = BLOCK / 64;
  hipLaunchKernelGGL(dense_score_kernel,
      dim3(ceil_div(B, rows_per_block)), dim3(BLOCK), 0, stream,
      x.data_ptr<float>(), medians.data_ptr<float>(), inv_scale.data_ptr<float>(),
      w.data_ptr<float>(), (float)bias, (float)z_threshold, B, F,
      proba.data_ptr<double>(), iscore.data_ptr<double>(), outlier.data_ptr<double>());
  HIP_CHECK(hipGetLastError());
  return {proba, iscore, outlier};
}

// Fused impute + transpose for the large-batch dense drift path:
// x [B, F] row-major -> xt [F, B] row-major with NaN -> median, one pass
// through an LDS tile (32x32, +1 padding column against bank conflicts).
__global__ __launch_bounds__(256) void impute_transpose_kernel(
    const float* __restrict__ x,   // [B, F]
    const float* __restrict__ medians,  // [F]
    int n_rows,
    int n_feat,
    float* __restrict__ xt)        // [F, B]
{
  __shared__ float tile[32][33];
  const int f0 = blockIdx.x * 32;
  const int r0 = blockIdx.y * 32;
  const int tx = threadIdx.x & 31;   // feature within tile on load
  const int ty = threadIdx.x >> 5;   // row group (8 rows per pass)
#pragma unroll
  for (int k = 0; k < 4; ++k) {
    const int r = r0 + ty + k * 8;
    const int f = f0 + tx;
    float v = 0.0f;
    if (r < n_rows && f < n_feat) {
      v = x[(size_t)r * n_feat + f];
      if (isnan(v)) v = medians[f];
    }
    tile[ty + k * 8][tx] = v;
  }
  __syncthreads();
#pragma unroll
  for (int k = 0; k < 4; ++k) {
    const int f = f0 + ty + k * 8;   // feature on store
    const int r = r0 + tx;           // row on store (coalesced)
    if (f < n_feat && r < n_rows) xt[(size_t)f * n_rows + r] = tile[tx][ty + k * 8];
  }
}

torch::Tensor impute_transpose(torch::Tensor x, torch::Tensor medians) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kFloat32 && x.is_contiguous());
  const int B = (int)x.size(0);
  const int F = (int)x.size(1);
  TORCH_CHECK((int)medians.numel() == F, "median size mismatch");
  auto xt = torch::empty({F, B},
      torch::TensorOptions().dtype(torch::kFloat32).device(x.device()));
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(impute_transpose_kernel,
      dim3(ceil_div(F, 32), ceil_div(B, 32)), dim3(256), 0, stream,
      x.data_ptr<float>(), medians.data_ptr<float>(), B, F,
      xt.data_ptr<float>());
  HIP_CHECK(hipGetLastError());
  return xt;
}

// Scan-only K-S for pre-sorted batch columns (large-B dense path: the sort
// goes through rocPRIM (torch.sort), then this kernel fills the chip with
// (feature x row-chunk) blocks; per-feature max via f32-bit atomicMax).
__global__ __launch_bounds__(BLOCK) void ks_scan_kernel(
    const float* __restrict__ xs,         // [F, B] row-contiguous, each row sorted
    int n_cols,
    int n_rows,
    const float* __restrict__ ref_sorted,
    const int64_t* __restrict__ rs_off,
    unsigned int* __restrict__ ks_bits)   // [F] f32 bits, pre-zeroed
{
  const int j = blockIdx.x;
  const int i = blockIdx.y * blockDim.x + threadIdx.x;
  const int m = n_rows;
  const int64_t lo = rs_off[j];
  const int n = (int)(rs_off[j + 1] - lo);
  const float* __restrict__ ref = ref_sorted + lo;
  const float* __restrict__ col = xs + (size_t)j * n_rows;  // contiguous

  double dmax = 0.0;
  if (i < m) {
    const float b = col[i];
    int l = 0, r = n;
    while (l < r) { const int mid = (l + r) >> 1; if (ref[mid] < b) l = mid + 1; else r = mid; }
    const int sl = l;
    r = n;
    while (l < r) { const int mid = (l + r) >> 1; if (ref[mid] <= b) l = mid + 1; else r = mid; }
    const int sr = l;
    int bl = 0; r = i;  // tie-run bounds within the sorted column
    while (bl < r) { const int mid = (bl + r) >> 1;
      if (col[mid] < b) bl = mid + 1; else r = mid; }
    int br = i + 1; r = m;
    while (br < r) { const int mid = (br + r) >> 1;
      if (col[mid] <= b) br = mid + 1; else r = mid; }
    const double fl = fabs((double)sl / n - (double)bl / m);
    const double fr = fabs((double)sr / n - (double)br / m);
    dmax = fmax(fl, fr);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    dmax = fmax(dmax, __shfl_down(dmax, off, 64));
  if ((threadIdx.x & 63) == 0 && dmax > 0.0)
    atomicMax(&ks_bits[j], __float_as_uint((float)dmax));
}

torch::Tensor ks_stats_sorted(
    torch::Tensor xs_sorted, torch::Tensor ref_sorted, torch::Tensor rs_off)
{
  TORCH_CHECK(xs_sorted.is_cuda() && xs_sorted.scalar_type() == torch::kFloat32
              && xs_sorted.is_contiguous());
  const int F = (int)xs_sorted.size(0);  // [F, B]: per-feature sorted rows
  const int B = (int)xs_sorted.size(1);
  TORCH_CHECK((int)rs_off.size(0) == F + 1, "rs_off size mismatch");
  auto bits = torch::zeros({F},
      torch::TensorOptions().dtype(torch::kInt32).device(xs_sorted.device()));
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  auto rs_off64 = rs_off.to(torch::kInt64);
  hipLaunchKernelGGL(ks_scan_kernel, dim3(F, ceil_div(B, BLOCK)), dim3(BLOCK), 0, stream,
      xs_sorted.data_ptr<float>(), F, B,
      ref_sorted.data_ptr<float>(), rs_off64.data_ptr<int64_t>(),
      reinterpret_cast<unsigned int*>(bits.data_ptr<int>()));
  HIP_CHECK(hipGetLastError());
  return bits.view(torch::kFloat32);
}

// Generic exact K-S D over any column count (dense drift path; the credit
// path embeds the same kernel in its session graph).
torch::Tensor ks_stats(
    torch::Tensor nums, torch::Tensor medians,
    torch::Tensor ref_sorted, torch::Tensor rs_off,
    int64_t block, int64_t ref_lds)
{
  TORCH_CHECK(nums.is_cuda() && nums.scalar_type() == torch::kFloat32 && nums.is_contiguous());
  const int B = (int)nums.size(0);
  const int F = (int)nums.size(1);
  TORCH_CHECK(B <= MAX_DRIFT_ROWS, "K-S batch too large: ", B);
  TORCH_CHECK((int)rs_off.size(0) == F + 1, "rs_off size mismatch");
  TORCH_CHECK(block == 256 || block == 512, "block must be 256 or 512");
  auto ks_d = torch::empty({F},
      torch::TensorOptions().dtype(torch::kFloat32).device(nums.device()));
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  int m_pow2 = (int)block / 64;  // >= one reduction slot per wave
  while (m_pow2 < B) m_pow2 <<= 1;
  size_t smem = (size_t)m_pow2 * sizeof(float);
  auto rs_off64 = rs_off.to(torch::kInt64);
  if (ref_lds) {
    int64_t maxr = 0;
    auto ro = rs_off64.cpu();
    auto* rp = ro.data_ptr<int64_t>();
    for (int j = 0; j < F; ++j) maxr = std::max<int64_t>(maxr, rp[j + 1] - rp[j]);
    smem += (size_t)maxr * sizeof(float);
    TORCH_CHECK(smem <= KS_LDS_BYTES_MAX, "ref too large for LDS");
  }
  if (block == 512)
    hipLaunchKernelGGL((ks_kernel_t<512>), dim3(F), dim3(512), smem, stream,
        nums.data_ptr<float>(), medians.data_ptr<float>(), F, B, m_pow2,
        (int)ref_lds, ref_sorted.data_ptr<float>(),
        rs_off64.data_ptr<int64_t>(), ks_d.data_ptr<float>());
  else
    hipLaunchKernelGGL((ks_kernel_t<256>), dim3(F), dim3(256), smem, stream,
        nums.data_ptr<float>(), medians.data_ptr<float>(), F, B, m_pow2,
        (int)ref_lds, ref_sorted.data_ptr<float>(),
        rs_off64.data_ptr<int64_t>(), ks_d.data_ptr<float>());
  HIP_CHECK(hipGetLastError());
  return ks_d;
}

// ---------------------------------------------------------------------------
// ScoreSession — one replica's resident GPU state + a single-call scoring
// path. Owns the model buffers in HBM, a private HIP stream, device
// workspace and pinned staging, so one Python call per request performs:
//   pinned H2D -> zero accumulators -> forest x2 -> finalize -> drift -> D2H
// with no per-call torch dispatch or allocation (the Python-side hot loop
// was ~10x the kernel time). The GIL is released while waiting.
// ---------------------------------------------------------------------------

// Completion waits: hipStreamSynchronize/hipEventSynchronize block on the
// runtime's interrupt-driven wakeup, which costs tens of µs of wake
// latency per request — a large share of the ~60 µs/request host floor at
// microsecond kernel times. Serving processes own their core, so the
// default is a polling wait (hipStreamQuery granularity ~1-2 µs);
// CREDITCORE_BLOCK_SYNC=1 restores blocking waits for A/B or
// share-the-host deployments.
inline bool spin_sync_enabled() {
  static const bool on = (std::getenv("CREDITCORE_BLOCK_SYNC") == nullptr);
  return on;
}

inline void sync_stream(hipStream_t s) {
  if (spin_sync_enabled()) {
    hipError_t e;
    while ((e = hipStreamQuery(s)) == hipErrorNotReady) {}
    if (e != hipSuccess) HIP_CHECK(e);
  } else {
    HIP_CHECK(hipStreamSynchronize(s));
  }
}

inline void sync_event(hipEvent_t ev) {
  // Event waits run on the EPILOGUE thread concurrently with the main
  // thread's hipGraphLaunch: a hipEventQuery spin there hammers the HIP
  // runtime's internal locks and was measured to double the launch cost
  // (submit 51 -> 96 us at b=1024). Block by default; CREDITCORE_SPIN_EVENT=1
  // restores the spin for single-threaded callers.
  static const bool spin = (std::getenv("CREDITCORE_SPIN_EVENT") != nullptr);
  if (spin) {
    hipError_t e;
    while ((e = hipEventQuery(ev)) == hipErrorNotReady) {}
    if (e != hipSuccess) HIP_CHECK(e);
  } else {
    HIP_CHECK(hipEventSynchronize(ev));
  }
}

struct ScoreSession {
  torch::Tensor cls_nodes, cls_off, feat_col, feat_code, medians;
  torch::Tensor if_nodes, if_off, ref_sorted, rs_off, cat_off;
  torch::Tensor d_codes, d_nums, acc, outs, d_drift, d_ks_scratch;
  size_t drift_bytes = 0, drift_ksd_off = 0;  // [hist i32 | ksd f32] blob
  torch::Tensor pin_codes, pin_nums, pin_outs, pin_drift;
  torch::Tensor pin_hist, pin_ksd;  // typed views into pin_drift
  double if_denom{}, if_offset{}, if_threshold{};
  int cls_kind{};
  double cls_bias{};
  int64_t total_bins{}, t_cls{}, t_if{}, capacity{};
  int device_index{};
  hipStream_t stream{};
  hipStream_t stream2{};  // drift branch (runs parallel to the forests)
  hipEvent_t ev_fork{}, ev_join{};
  hipEvent_t ev_done[2]{};  // per-slot completion (async score)

  // raw slot pointers into the pinned buffers
  int16_t* p_codes(int s) { return pin_codes.data_ptr<int16_t>() + (size_t)s * capacity * N_CAT; }
  float* p_nums(int s) { return pin_nums.data_ptr<float>() + (size_t)s * capacity * N_NUM; }
  double* p_outs(int s) { return pin_outs.data_ptr<double>() + (size_t)s * 3 * capacity; }
  uint8_t* p_drift(int s) { return pin_drift.data_ptr<uint8_t>() + (size_t)s * drift_bytes; }
  int32_t* p_hist(int s) { return reinterpret_cast<int32_t*>(p_drift(s)); }
  float* p_ksd(int s) { return reinterpret_cast<float*>(p_drift(s) + drift_ksd_off); }
  int32_t* d_hist() { return reinterpret_cast<int32_t*>(d_drift.data_ptr<uint8_t>()); }
  float* d_ksd() { return reinterpret_cast<float*>(d_drift.data_ptr<uint8_t>() + drift_ksd_off); }

  ScoreSession(py::dict model, int64_t cap, int dev) : capacity(cap), device_index(dev) {
    c10::hip::HIPGuard guard((c10::DeviceIndex)dev);
    auto devopt = torch::TensorOptions().device(torch::kCUDA, dev);
    auto up_i32 = [&](const char* k) {
      return py::cast<torch::Tensor>(model[k]).to(devopt.dtype(torch::kInt32)).contiguous();
    };
    auto up_f32 = [&](const char* k) {
      return py::cast<torch::Tensor>(model[k]).to(devopt.dtype(torch::kFloat32)).contiguous();
    };
    cls_nodes = up_i32("cls_nodes");
    cls_off = up_i32("cls_tree_offsets");
    feat_col = up_i32("feat_col");
    feat_code = up_i32("feat_code");
    medians = up_f32("medians");
    if_nodes = up_i32("if_nodes");
    if_off = up_i32("if_tree_offsets");
    ref_sorted = up_f32("ref_sorted");
    rs_off = py::cast<torch::Tensor>(model["ref_sorted_offsets"])
                 .to(devopt.dtype(torch::kInt64)).contiguous();
    cat_off = up_i32("ref_cat_offsets");
    if (model.contains("cls_kind")) cls_kind = py::cast<int>(model["cls_kind"]);
    if (model.contains("cls_bias")) cls_bias = py::cast<double>(model["cls_bias"]);
    if_denom = py::cast<double>(model["if_denom"]);
    if_offset = py::cast<double>(model["if_offset"]);
    if_threshold = py::cast<double>(model["if_threshold"]);
    t_cls = cls_off.size(0) - 1;
    t_if = if_off.size(0) - 1;
    total_bins = py::cast<torch::Tensor>(model["ref_cat_offsets"])[-1].item<int64_t>();

    d_codes = torch::empty({capacity, N_CAT}, devopt.dtype(torch::kInt16));
    d_nums = torch::empty({capacity, N_NUM}, devopt.dtype(torch::kFloat32));
    // all-zero invariant: finalize_kernel re-zeros the rows it consumes,
    // so record() never needs a memset node for the accumulators
    acc = torch::zeros({2, capacity}, devopt.dtype(torch::kFloat64));
    outs = torch::empty({3, capacity}, devopt.dtype(torch::kFloat64));
    // drift outputs packed into one blob: [hist i32 | ksd f32] — both are
    // b-independent sizes, so one D2H covers the whole drift branch
    drift_ksd_off = (size_t)total_bins * sizeof(int32_t);
    drift_bytes = drift_ksd_off + (size_t)N_NUM * sizeof(float);
    d_drift = torch::empty({(int64_t)drift_bytes}, devopt.dtype(torch::kUInt8));
    // per-(column, block) partial maxima for the multi-block counting K-S
    d_ks_scratch = torch::empty({(int64_t)N_NUM * KS_MB_MAX_G},
                                devopt.dtype(torch::kFloat32));

    // two slots: step i's host epilogue reads slot i%2 while step i+1's
    // graph fills the other slot (pipelined serving/bench loops)
    auto pinned = torch::TensorOptions().pinned_memory(true);
    pin_codes = torch::empty({2, capacity, N_CAT}, pinned.dtype(torch::kInt16));
    pin_nums = torch::empty({2, capacity, N_NUM}, pinned.dtype(torch::kFloat32));
    pin_outs = torch::empty({2, 3, capacity}, pinned.dtype(torch::kFloat64));
    pin_drift = torch::empty({2, (int64_t)drift_bytes}, pinned.dtype(torch::kUInt8));
    pin_hist = pin_drift.slice(1, 0, (int64_t)drift_ksd_off).view(torch::kInt32);
    pin_ksd = pin_drift.slice(1, (int64_t)drift_ksd_off, (int64_t)drift_bytes)
                  .view(torch::kFloat32);

    auto rs = py::cast<torch::Tensor>(model["ref_sorted_offsets"]);
    auto rs_acc = rs.accessor<int32_t, 1>();
    for (int j = 0; j + 1 < rs.size(0); ++j)
      max_ref_len = std::max(max_ref_len, (int64_t)(rs_acc[j + 1] - rs_acc[j]));

    // opt in to >64 KiB dynamic LDS for every launched K-S instantiation:
    // record() launches ks_kernel_t<512> (b=16384 needs exactly 64 KiB of
    // dynamic LDS, at/over the historical default cap), and the standalone
    // drift_stats/ks_stats paths launch <256>.
    (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&ks_kernel_t<256>),
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)KS_LDS_BYTES);
    (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&ks_kernel_t<512>),
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)KS_LDS_BYTES);
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&ks_count_mb_kernel_t<512>),
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)KS_LDS_BYTES);

    HIP_CHECK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
    HIP_CHECK(hipStreamCreateWithFlags(&stream2, hipStreamNonBlocking));
    HIP_CHECK(hipEventCreateWithFlags(&ev_fork, hipEventDisableTiming));
    HIP_CHECK(hipEventCreateWithFlags(&ev_join, hipEventDisableTiming));
    HIP_CHECK(hipEventCreateWithFlags(&ev_done[0], hipEventDisableTiming));
    HIP_CHECK(hipEventCreateWithFlags(&ev_done[1], hipEventDisableTiming));
    HIP_CHECK(hipDeviceSynchronize());  // uploads above used torch's stream
  }

  ~ScoreSession() {
    if (stream) {
      (void)hipStreamSynchronize(stream);
      for (auto& kv : graphs) (void)hipGraphExecDestroy(kv.second);
      (void)hipStreamDestroy(stream);
      (void)hipStreamDestroy(stream2);
      (void)hipEventDestroy(ev_fork);
      (void)hipEventDestroy(ev_join);
      (void)hipEventDestroy(ev_done[0]);
      (void)hipEventDestroy(ev_done[1]);
    }
  }

  // Record the full scoring sequence for batch size b on `stream`.
  // Output layout (b-packed so one D2H covers all three): outs holds
  // proba[0:b] | iscore[b:2b] | outlier[2b:3b]; pin_outs mirrors it.
  //
  // Two-stream shape (round-1 tuned): the drift branch (histogram + K-S)
  // forks onto stream2 off the H2D copies and joins after the output
  // copy. Both forests run as ONE dual-grid launch (forest_kernel_dual)
  // so the iforest rides along with the classifier instead of
  // serializing after it; a 3-stream fork for the same effect was
  // measured SLOWER (extra cross-stream graph edges ≈ +20 µs/replay at
  // b=1024 — kernel_tuning.md).
  void record(int b, bool with_drift, int slot) {
    HIP_CHECK(hipMemcpyAsync(d_codes.data_ptr(), p_codes(slot),
        (size_t)b * N_CAT * sizeof(short), hipMemcpyHostToDevice, stream));
    HIP_CHECK(hipMemcpyAsync(d_nums.data_ptr(), p_nums(slot),
        (size_t)b * N_NUM * sizeof(float), hipMemcpyHostToDevice, stream));
    // fork point: the drift branch (stream2) depends only on the H2D
    // copies. At tiny batches the drift kernels are microseconds and the
    // fork/join event edges cost more graph-replay overhead than the
    // overlap saves — run the branch inline on `stream` instead.
    const bool fork_drift = with_drift && b > 64;
    hipStream_t sdrift = fork_drift ? stream2 : stream;
    if (fork_drift) {
      HIP_CHECK(hipEventRecord(ev_fork, stream));
      HIP_CHECK(hipStreamWaitEvent(stream2, ev_fork, 0));
    }
    double* acc_cls = acc.data_ptr<double>();
    double* acc_if = acc_cls + b;  // b-packed; all-zero by invariant

    const int row_blocks = ceil_div(b, BLOCK);
    // 2-tree-ILP traversal measured faster at every batch size
    // (bench/kernel_micro.py: -12% @1k rows, -42% @16k); grid-y covers
    // ceil(T/2) chunks, each thread walking trees t and t+n_chunks.
    auto chunks = [&](int64_t t) {
      return std::max(1, std::min(ceil_div(2048, row_blocks), (int)((t + 1) / 2)));
    };
    const int c_cls = chunks(t_cls);
    const int c_if = chunks(t_if);
    hipLaunchKernelGGL(forest_kernel_dual, dim3(row_blocks, c_cls + c_if),
        dim3(BLOCK), 0, stream,
        d_codes.data_ptr<short>(), d_nums.data_ptr<float>(), medians.data_ptr<float>(),
        reinterpret_cast<const int4*>(cls_nodes.data_ptr<int>()),
        cls_off.data_ptr<int>(), (int)t_cls,
        feat_col.data_ptr<int>(), feat_code.data_ptr<int>(),
        reinterpret_cast<const int4*>(if_nodes.data_ptr<int>()),
        if_off.data_ptr<int>(), (int)t_if, c_cls, b, acc_cls, acc_if);

    double* proba = outs.data_ptr<double>();
    hipLaunchKernelGGL(finalize_kernel, dim3(row_blocks), dim3(BLOCK), 0, stream,
        acc_cls, acc_if, b, cls_kind, 1.0 / (double)t_cls, cls_bias,
        if_denom, if_offset, if_threshold,
        proba, proba + b, proba + 2 * b);

    if (with_drift) {
      // Drift branch: when forked (b > 64) the K-S and categorical
      // histogram run on stream2 in parallel with the forest chain
      // (captured as parallel graph branches, joined after the output
      // copy); at tiny batches everything stays serial on `stream`.
      // small batches: one block overwrites the histogram (no memset node);
      // larger ones pre-zero + atomically accumulate across blocks
      const int hist_blocks = (b <= 2048) ? 1 : std::min(row_blocks, 1024);
      if (hist_blocks > 1)
        HIP_CHECK(hipMemsetAsync(d_hist(), 0, (size_t)total_bins * sizeof(int), sdrift));
      hipLaunchKernelGGL(cat_hist_kernel, dim3(hist_blocks), dim3(BLOCK),
          (size_t)total_bins * sizeof(int), sdrift,
          d_codes.data_ptr<short>(), b, cat_off.data_ptr<int>(), (int)total_bins,
          d_hist());
      if (b <= KS_COUNT_MAX_ROWS) {
        // O(B^2) counting path: no sort, no barrier chain (38.6 -> single-
        // digit us at b=1024; see profiles/kernel_tuning.md)
        const size_t smem = (size_t)std::max(b, 512 / 64) * sizeof(float);
        hipLaunchKernelGGL((ks_count_kernel_t<512>), dim3(N_NUM), dim3(512),
            smem, sdrift,
            d_nums.data_ptr<float>(), medians.data_ptr<float>(), N_NUM, b,
            ref_sorted.data_ptr<float>(), rs_off.data_ptr<int64_t>(), d_ksd());
      } else if (std::getenv("CREDITCORE_KS_BITONIC") != nullptr) {
        // round-1 path kept for A/B: one bitonic-sort block per column
        // (only n_cols of 256 CUs busy; 38.6 us at b=1024, 509 at 16k)
        int m_pow2 = 512 / 64;  // >= one cross-wave reduction slot per wave
        while (m_pow2 < b) m_pow2 <<= 1;
        // measured (bench/kernel_micro.py on MI355X): staging the ref
        // column in LDS is slower at b=1024 (62 vs 48 us) and within noise
        // at 16k, so the ref stays in L2 (ref_lds=0).
        // 512 threads: halves the bitonic's serial depth per thread
        // (kernel_micro: 48.6->31.2 us @1k, 963->509 @16k)
        hipLaunchKernelGGL((ks_kernel_t<512>), dim3(N_NUM), dim3(512),
            (size_t)m_pow2 * sizeof(float), sdrift,
            d_nums.data_ptr<float>(), medians.data_ptr<float>(), N_NUM, b,
            m_pow2, /*ref_lds=*/0, ref_sorted.data_ptr<float>(),
            rs_off.data_ptr<int64_t>(), d_ksd());
      } else {
        // multi-block counting path: n_cols*g workgroups fill the chip
        // (the sort/count single-block kernels used 14 of 256 CUs)
        const int g = ks_mb_blocks(b);
        const size_t smem = (size_t)std::max(b, 512 / 64) * sizeof(float);
        hipLaunchKernelGGL((ks_count_mb_kernel_t<512>), dim3(N_NUM, g),
            dim3(512), smem, sdrift,
            d_nums.data_ptr<float>(), medians.data_ptr<float>(), N_NUM, b,
            ref_sorted.data_ptr<float>(), rs_off.data_ptr<int64_t>(),
            d_ks_scratch.data_ptr<float>());
        hipLaunchKernelGGL(ks_count_reduce_kernel, dim3(1), dim3(64), 0,
            sdrift, d_ks_scratch.data_ptr<float>(), g, N_NUM, d_ksd());
      }
      // one D2H for the whole drift branch (hist + K-S D share a blob)
      HIP_CHECK(hipMemcpyAsync(p_drift(slot), d_drift.data_ptr<uint8_t>(),
          drift_bytes, hipMemcpyDeviceToHost, sdrift));
      if (fork_drift) HIP_CHECK(hipEventRecord(ev_join, sdrift));
    }
    // classifier-output D2H depends only on finalize — it overlaps the
    // drift branch's K-S tail; the join lands after it so graph completion
    // still covers both streams
    HIP_CHECK(hipMemcpyAsync(p_outs(slot), proba,
        (size_t)(3 * b) * sizeof(double), hipMemcpyDeviceToHost, stream));
    if (fork_drift) HIP_CHECK(hipStreamWaitEvent(stream, ev_join, 0));
    HIP_CHECK(hipGetLastError());
  }

  // Score b rows already staged in pin_codes/pin_nums. Blocks (GIL
  // released) until pin_outs/pin_hist/pin_ksd hold the results. The whole
  // sequence is captured into a hipGraph per (b, with_drift) and replayed
  // as one submit on subsequent requests of the same shape.
  void score(int64_t b64, bool with_drift, bool sync, int64_t slot64) {
    TORCH_CHECK(b64 >= 1 && b64 <= capacity, "batch out of range: ", b64);
    TORCH_CHECK(!with_drift || b64 <= MAX_DRIFT_ROWS, "drift batch too large: ", b64);
    TORCH_CHECK(slot64 == 0 || slot64 == 1, "slot must be 0/1");
    const int b = (int)b64;
    const int slot = (int)slot64;
    py::gil_scoped_release nogil;
    c10::hip::HIPGuard guard((c10::DeviceIndex)device_index);

    const uint64_t key = ((uint64_t)b << 2) | ((with_drift ? 1u : 0u) << 1) | (uint64_t)slot;
    // Only capture graphs for recurring shapes (powers of two and
    // 256-multiples — what the micro-batcher and bench produce). Arbitrary
    // merged sizes run the eager path directly: a capture costs ~ms, an
    // eager pass costs ~30 µs extra — capture churn would swamp the win.
    const bool graphable = ((b & (b - 1)) == 0) || (b % 256 == 0);
    auto it = graphs.find(key);
    if (it == graphs.end()) {
      if (!graphable || graphs.size() >= 128) {
        record(b, with_drift, slot);
        if (sync) sync_stream(stream);
        else HIP_CHECK(hipEventRecord(ev_done[slot], stream));
        return;
      }
      hipGraph_t graph;
      HIP_CHECK(hipStreamBeginCapture(stream, hipStreamCaptureModeThreadLocal));
      record(b, with_drift, slot);
      HIP_CHECK(hipStreamEndCapture(stream, &graph));
      hipGraphExec_t exec;
      HIP_CHECK(hipGraphInstantiate(&exec, graph, nullptr, nullptr, 0));
      HIP_CHECK(hipGraphDestroy(graph));
      it = graphs.emplace(key, exec).first;
    }
    HIP_CHECK(hipGraphLaunch(it->second, stream));
    if (sync) sync_stream(stream);
    else HIP_CHECK(hipEventRecord(ev_done[slot], stream));
  }

  void wait_slot(int64_t slot) {
    py::gil_scoped_release nogil;
    sync_event(ev_done[slot & 1]);
  }

  std::unordered_map<uint64_t, hipGraphExec_t> graphs;
  int64_t max_ref_len{};
  static constexpr size_t KS_LDS_BYTES = KS_LDS_BYTES_MAX;

  void synchronize() {
    py::gil_scoped_release nogil;
    HIP_CHECK(hipStreamSynchronize(stream));
  }
};

// ---------------------------------------------------------------------------
// Host-side request encoder (the native data-loader for the serving path).
// Replaces the reference's pandas DataFrame construction + sklearn
// OneHotEncoder lookup per request (reference app/main.py:54 →
// ColumnTransformer at 01-train cell-6) with one C pass over the parsed
// request dicts: ~20 µs per 1024-row request vs ~2 ms in Python.
// ---------------------------------------------------------------------------

#include <pybind11/numpy.h>

#include <cmath>
#include <string>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

py::tuple encode_records(py::list recs, py::list vocabs, py::list cat_names,
                         py::list num_names, py::str missing_cat) {
  const ssize_t b = py::len(recs);
  const int ncat = (int)py::len(cat_names);
  const int nnum = (int)py::len(num_names);
  py::array_t<int16_t> codes({b, (ssize_t)ncat});
  py::array_t<float> nums({b, (ssize_t)nnum});
  auto* cp = codes.mutable_data();
  auto* fp = nums.mutable_data();

  std::vector<std::unordered_map<std::string, int16_t>> maps(ncat);
  std::vector<int16_t> missing_code(ncat, -1);
  const std::string miss = py::cast<std::string>(missing_cat);
  for (int j = 0; j < ncat; ++j) {
    py::list v = vocabs[j];
    for (ssize_t k = 0; k < py::len(v); ++k)
      maps[j][py::cast<std::string>(v[k])] = (int16_t)k;
    auto it = maps[j].find(miss);
    if (it != maps[j].end()) missing_code[j] = it->second;
  }
  std::vector<PyObject*> ckeys(ncat), nkeys(nnum);  // borrowed refs
  for (int j = 0; j < ncat; ++j) ckeys[j] = PyList_GET_ITEM(cat_names.ptr(), (ssize_t)j);
  for (int j = 0; j < nnum; ++j) nkeys[j] = PyList_GET_ITEM(num_names.ptr(), (ssize_t)j);

  for (ssize_t i = 0; i < b; ++i) {
    PyObject* r = PyList_GET_ITEM(recs.ptr(), i);
    if (!PyDict_Check(r)) throw py::type_error("record must be a dict");
    for (int j = 0; j < ncat; ++j) {
      PyObject* v = PyDict_GetItem(r, ckeys[j]);  // borrowed, NULL if absent
      int16_t code;
      if (v == nullptr || v == Py_None) {
        code = missing_code[j];  // SimpleImputer fill_value="missing"
      } else if (PyUnicode_Check(v)) {
        Py_ssize_t len;
        const char* s = PyUnicode_AsUTF8AndSize(v, &len);
        auto it = maps[j].find(std::string(s, (size_t)len));
        // unknown category -> -1 (OneHotEncoder handle_unknown="ignore")
        code = (it == maps[j].end()) ? (int16_t)-1 : it->second;
      } else if (PyFloat_Check(v) && std::isnan(PyFloat_AS_DOUBLE(v))) {
        code = missing_code[j];  // pandas-style NaN missing marker
      } else {
        throw py::value_error("categorical field must be a string/None");
      }
      cp[i * ncat + j] = code;
    }
    for (int j = 0; j < nnum; ++j) {
      PyObject* v = PyDict_GetItem(r, nkeys[j]);
      float x;
      if (v == nullptr || v == Py_None) {
        x = NAN;  // imputed to the median inside the scoring kernel
      } else if (PyFloat_Check(v)) {
        x = (float)PyFloat_AS_DOUBLE(v);
      } else if (PyLong_Check(v)) {
        x = (float)PyLong_AsDouble(v);
      } else {
        throw py::value_error("numeric field must be a number/None");
      }
      fp[i * nnum + j] = x;
    }
  }
  return py::make_tuple(codes, nums);
}

// ---------------------------------------------------------------------------
// encode_json — parse a /score request body (JSON bytes, the wire format:
// [{"sex": "male", ..., "credit_limit": 18000.0, ...}, ...]) straight into
// (codes, nums) with no intermediate Python objects. This is the serving
// fast path; any deviation from the expected shape throws and the caller
// falls back to the pydantic path for a proper 422. The GIL is released
// during the parse.
// ---------------------------------------------------------------------------

#include <charconv>
#include <cstring>

namespace jsonenc {

struct Error {
  std::string msg;
};

struct Parser {
  const char* p;
  const char* end;

  [[noreturn]] void fail(const char* what) {
    throw Error{std::string(what) + " at offset " + std::to_string((size_t)(p - begin_))};
  }
  const char* begin_;

  void ws() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) ++p;
  }
  bool eat(char c) {
    ws();
    if (p < end && *p == c) { ++p; return true; }
    return false;
  }
  void expect(char c, const char* what) {
    if (!eat(c)) fail(what);
  }

  // Parse a JSON string starting at the opening quote; returns the raw
  // span between quotes and whether it contains escapes. Fast path: one
  // SIMD memchr to the closing quote + one over the span for backslashes
  // (escapes are rare in this schema's values).
  void raw_string(const char*& s, size_t& len, bool& escaped) {
    expect('"', "expected string");
    s = p;
    const char* q = (const char*)std::memchr(p, '"', (size_t)(end - p));
    if (q == nullptr) { p = end; fail("unterminated string"); }
    if (std::memchr(p, '\\', (size_t)(q - p)) == nullptr) {
      escaped = false;
      len = (size_t)(q - s);
      p = q + 1;
      return;
    }
    escaped = true;
    while (p < end) {
      const char c = *p;
      if (c == '"') { len = (size_t)(p - s); ++p; return; }
      if (c == '\\') {
        ++p;
        if (p >= end) break;
      }
      ++p;
    }
    fail("unterminated string");
  }

  // Unescape into buf (capacity cap); returns length or SIZE_MAX if too long.
  size_t unescape(const char* s, size_t len, char* buf, size_t cap) {
    size_t o = 0;
    for (size_t i = 0; i < len; ++i) {
      char c = s[i];
      if (c != '\\') {
        if (o >= cap) return SIZE_MAX;
        buf[o++] = c;
        continue;
      }
      if (++i >= len) return SIZE_MAX;
      c = s[i];
      char repl;
      switch (c) {
        case '"': repl = '"'; break;
        case '\\': repl = '\\'; break;
        case '/': repl = '/'; break;
        case 'b': repl = '\b'; break;
        case 'f': repl = '\f'; break;
        case 'n': repl = '\n'; break;
        case 'r': repl = '\r'; break;
        case 't': repl = '\t'; break;
        case 'u': {
          if (i + 4 >= len) return SIZE_MAX;
          unsigned cp = 0;
          for (int k = 1; k <= 4; ++k) {
            const char h = s[i + k];
            cp <<= 4;
            if (h >= '0' && h <= '9') cp |= (unsigned)(h - '0');
            else if (h >= 'a' && h <= 'f') cp |= (unsigned)(h - 'a' + 10);
            else if (h >= 'A' && h <= 'F') cp |= (unsigned)(h - 'A' + 10);
            else return SIZE_MAX;
          }
          i += 4;
          // UTF-8 encode (surrogate pairs unsupported here: vocab is ASCII,
          // a non-matching value only needs to be *skipped* correctly)
          if (cp < 0x80) {
            if (o >= cap) return SIZE_MAX;
            buf[o++] = (char)cp;
          } else if (cp < 0x800) {
            if (o + 2 > cap) return SIZE_MAX;
            buf[o++] = (char)(0xC0 | (cp >> 6));
            buf[o++] = (char)(0x80 | (cp & 0x3F));
          } else {
            if (o + 3 > cap) return SIZE_MAX;
            buf[o++] = (char)(0xE0 | (cp >> 12));
            buf[o++] = (char)(0x80 | ((cp >> 6) & 0x3F));
            buf[o++] = (char)(0x80 | (cp & 0x3F));
          }
          continue;
        }
        default: return SIZE_MAX;
      }
      if (o >= cap) return SIZE_MAX;
      buf[o++] = repl;
    }
    return o;
  }

  double number() {
    ws();
    double v;
    auto [q, ec] = std::from_chars(p, end, v);
    if (ec != std::errc()) fail("expected number");
    p = q;
    return v;
  }

  bool literal(const char* lit) {
    const size_t n = std::strlen(lit);
    if ((size_t)(end - p) >= n && std::memcmp(p, lit, n) == 0) { p += n; return true; }
    return false;
  }

  void skip_value() {
    ws();
    if (p >= end) fail("truncated value");
    const char c = *p;
    if (c == '"') {
      const char* s; size_t l; bool e;
      raw_string(s, l, e);
    } else if (c == '{' || c == '[') {
      const char open = c, close = (c == '{') ? '}' : ']';
      int depth = 0;
      while (p < end) {
        const char d = *p;
        if (d == '"') { const char* s; size_t l; bool e; raw_string(s, l, e); continue; }
        if (d == open) ++depth;
        else if (d == close && --depth == 0) { ++p; return; }
        ++p;
      }
      fail("unterminated container");
    } else if (literal("null") || literal("true") || literal("false")) {
    } else {
      number();
    }
  }
};

}  // namespace jsonenc

// Persistent encoder state: column dispatch table keyed by
// (key_len << 8) | last_char — one int hash + at most a couple of memcmp
// per field instead of hashing 15-18-char strings 23k times per request.
struct JsonEncoderState {
  std::vector<std::string> key_store;
  std::unordered_map<uint32_t, std::vector<std::pair<int, int>>> dispatch;
  // dispatch value: (key_store index, column id)
  std::vector<std::vector<std::string>> vocab;
  std::vector<int16_t> missing_code;
  std::vector<int16_t> def_codes;
  std::vector<float> def_nums;
  int ncat{}, nnum{};

  JsonEncoderState(py::list vocabs, py::list cat_names, py::list num_names,
                   py::str missing_cat, py::array_t<int16_t> default_codes,
                   py::array_t<float> default_nums) {
    ncat = (int)py::len(cat_names);
    nnum = (int)py::len(num_names);
    vocab.resize(ncat);
    missing_code.assign(ncat, -1);
    const std::string miss = py::cast<std::string>(missing_cat);
    for (int j = 0; j < ncat; ++j) {
      key_store.push_back(py::cast<std::string>(cat_names[j]));
      py::list v = vocabs[j];
      for (ssize_t k = 0; k < py::len(v); ++k) {
        vocab[j].push_back(py::cast<std::string>(v[k]));
        if (vocab[j].back() == miss) missing_code[j] = (int16_t)k;
      }
    }
    for (int j = 0; j < nnum; ++j)
      key_store.push_back(py::cast<std::string>(num_names[j]));
    for (int j = 0; j < ncat + nnum; ++j) {
      const std::string& k = key_store[j];
      const uint32_t h = ((uint32_t)k.size() << 8) | (uint8_t)k.back();
      dispatch[h].emplace_back(j, j);
    }
    TORCH_CHECK((int)default_codes.size() == ncat && (int)default_nums.size() == nnum,
        "default row size mismatch");
    def_codes.assign(default_codes.data(), default_codes.data() + ncat);
    def_nums.assign(default_nums.data(), default_nums.data() + nnum);
  }

  inline int lookup(const char* ks, size_t kl) const {
    if (kl == 0) return -1;
    const uint32_t h = ((uint32_t)kl << 8) | (uint8_t)ks[kl - 1];
    auto it = dispatch.find(h);
    if (it == dispatch.end()) return -1;
    for (const auto& cand : it->second) {
      const std::string& name = key_store[cand.first];
      if (name.size() == kl && std::memcmp(name.data(), ks, kl) == 0)
        return cand.second;
    }
    return -1;
  }
};

// Parse the full body into flat row-major code/num vectors. Caller must
// NOT hold the GIL. Throws jsonenc::Error on malformed input.
static size_t parse_records_nogil(const JsonEncoderState& st, char* data,
                                  ssize_t blen, std::vector<int16_t>& codes,
                                  std::vector<float>& nums) {
  const int ncat = st.ncat;
  const int nnum = st.nnum;
  const auto& vocab = st.vocab;
  const int16_t* def_codes = st.def_codes.data();
  const float* def_nums = st.def_nums.data();
  size_t b = 0;
  {
    jsonenc::Parser P{data, data + blen};
    P.begin_ = data;
    codes.reserve((size_t)blen / 70 * ncat + ncat);
    nums.reserve((size_t)blen / 70 * nnum + nnum);
    {
      P.expect('[', "body must be a JSON array");
      if (!P.eat(']')) {
        do {
          P.expect('{', "record must be an object");
          codes.resize(codes.size() + ncat);
          nums.resize(nums.size() + nnum);
          int16_t* crow = codes.data() + b * ncat;
          float* nrow = nums.data() + b * nnum;
          // absent fields take the schema defaults (pydantic semantics,
          // reference app/model.py:8-34)
          std::memcpy(crow, def_codes, ncat * sizeof(int16_t));
          std::memcpy(nrow, def_nums, nnum * sizeof(float));
          // clients overwhelmingly emit fields in schema order (pydantic
          // model dumps, the sample request, pandas records): try the
          // expected next column with one memcmp before the hash lookup
          int expect_col = 0;
          if (!P.eat('}')) {
            do {
              const char* ks; size_t kl; bool kesc;
              P.raw_string(ks, kl, kesc);
              char kbuf[64];
              if (kesc) {
                const size_t n = P.unescape(ks, kl, kbuf, sizeof(kbuf));
                if (n == SIZE_MAX) P.fail("bad key");
                ks = kbuf; kl = n;
              }
              P.expect(':', "expected ':'");
              int col = -1;
              if (expect_col < ncat + nnum) {
                const std::string& exp = st.key_store[expect_col];
                if (exp.size() == kl && std::memcmp(exp.data(), ks, kl) == 0)
                  col = expect_col;
              }
              if (col < 0) col = st.lookup(ks, kl);
              expect_col = (col >= 0) ? col + 1 : expect_col;
              if (col < 0) {
                P.skip_value();  // extra fields ignored (schema extra="ignore")
              } else if (col < ncat) {
                const int j = col;
                P.ws();
                if (P.p < P.end && *P.p == 'n') {
                  P.fail("null not accepted for a string field");
                } else {
                  const char* vs; size_t vl; bool vesc;
                  P.raw_string(vs, vl, vesc);
                  char vbuf[64];
                  if (vesc) {
                    const size_t n = P.unescape(vs, vl, vbuf, sizeof(vbuf));
                    if (n == SIZE_MAX) { crow[j] = -1; continue; }
                    vs = vbuf; vl = n;
                  }
                  int16_t code = -1;  // unknown category -> all-zero one-hot
                  for (size_t k = 0; k < vocab[j].size(); ++k) {
                    const std::string& cand = vocab[j][k];
                    if (cand.size() == vl && std::memcmp(cand.data(), vs, vl) == 0) {
                      code = (int16_t)k;
                      break;
                    }
                  }
                  crow[j] = code;
                }
              } else {
                const int j = col - ncat;
                P.ws();
                if (P.p < P.end && *P.p == 'n') {
                  P.fail("null not accepted for a numeric field");
                } else {
                  nrow[j] = (float)P.number();
                }
              }
            } while (P.eat(','));
            P.expect('}', "expected '}'");
          }
          ++b;
        } while (P.eat(','));
        P.expect(']', "expected ']'");
      }
      P.ws();
      if (P.p != P.end) P.fail("trailing data");
    }
  }
  return b;
}

py::tuple encode_json_impl(const JsonEncoderState& st, py::bytes body) {
  const int ncat = st.ncat;
  const int nnum = st.nnum;
  char* data;
  ssize_t blen;
  if (PyBytes_AsStringAndSize(body.ptr(), &data, &blen) != 0)
    throw py::value_error("body must be bytes");
  std::vector<int16_t> codes;
  std::vector<float> nums;
  size_t b;
  {
    py::gil_scoped_release nogil;
    try {
      b = parse_records_nogil(st, data, blen, codes, nums);
    } catch (const jsonenc::Error& e) {
      py::gil_scoped_acquire gil;
      throw py::value_error(e.msg);
    }
  }

  py::array_t<int16_t> codes_arr({(ssize_t)b, (ssize_t)ncat});
  py::array_t<float> nums_arr({(ssize_t)b, (ssize_t)nnum});
  if (b) {
    std::memcpy(codes_arr.mutable_data(), codes.data(), b * ncat * sizeof(int16_t));
    std::memcpy(nums_arr.mutable_data(), nums.data(), b * nnum * sizeof(float));
  }
  return py::make_tuple(codes_arr, nums_arr);
}

// ---------------------------------------------------------------------------
// Drift p-value epilogue on host, in C: chi-square survival via the exact
// closed forms for integer dof (even: Poisson tail sum; odd: erfc +
// half-integer-gamma series, both from the Q(k+2) = Q(k) + term recurrence)
// and the Pelz-Good series for the two-sample K-S p-value (mirrors
// creditcore.models.drift._pelz_good_sf; validated against scipy in tests).
// Replaces ~0.4 ms of numpy/scipy per request with ~5 µs.
// ---------------------------------------------------------------------------

static double chi2_sf_int_dof(double x, int dof) {
  if (dof <= 0) return 1.0;
  if (x <= 0.0) return 1.0;
  // Q(1) = erfc(sqrt(x/2)); Q(2) = exp(-x/2);
  // Q(k+2) = Q(k) + (x/2)^(k/2) e^(-x/2) / Gamma(k/2 + 1)
  const double h = 0.5 * x;
  double q, term;
  int k;
  if (dof % 2 == 1) {
    q = std::erfc(std::sqrt(h));
    term = std::sqrt(h / M_PI) * std::exp(-h) * 2.0;  // k=1 increment
    k = 1;
  } else {
    q = std::exp(-h);
    term = h * std::exp(-h);  // k=2 increment
    k = 2;
  }
  while (k + 2 <= dof) {
    q += term;
    k += 2;
    term *= h / (0.5 * k);
  }
  return std::min(1.0, std::max(0.0, q));
}

// Exact P(D_n > d) for small n via the Marsaglia-Tsang-Wang (2003) matrix
// method (the same algorithm scipy's exact path uses for n <= 140). k-d
// band is small in practice (k ~ ceil(n d) ~ sqrt(n) near the null); the
// sf underflows to 0 for n d^2 > 18 long before m gets large.
static double mtw_sf(int n, double d) {
  if (d <= 0.0) return 1.0;
  if (d >= 1.0) return 0.0;
  if (n * d * d > 18.0) return 0.0;  // sf < ~2e-16
  const int k = (int)std::ceil(n * d);
  const double h = k - n * d;
  const int m = 2 * k - 1;
  std::vector<double> H((size_t)m * m, 0.0), Q((size_t)m * m, 0.0), tmp((size_t)m * m);
  for (int i = 0; i < m; ++i)
    for (int j = 0; j < m; ++j)
      if (i - j + 1 >= 0) H[(size_t)i * m + j] = 1.0;
  for (int i = 0; i < m; ++i) {
    H[(size_t)i * m] -= std::pow(h, i + 1);
    H[(size_t)(m - 1) * m + i] -= std::pow(h, m - i);
  }
  H[(size_t)(m - 1) * m] += (2 * h - 1 > 0 ? std::pow(2 * h - 1, m) : 0.0);
  for (int i = 0; i < m; ++i)
    for (int j = 0; j < m; ++j)
      if (i - j + 1 > 0)
        for (int g = 1; g <= i - j + 1; ++g) H[(size_t)i * m + j] /= g;
  // Q = H^n with power-of-two scaling to avoid under/overflow
  int eH = 0, eQ = 0;
  auto matmul = [&](const std::vector<double>& a, const std::vector<double>& b,
                    std::vector<double>& c) {
    for (int i = 0; i < m; ++i)
      for (int j = 0; j < m; ++j) {
        double s = 0.0;
        for (int g = 0; g < m; ++g) s += a[(size_t)i * m + g] * b[(size_t)g * m + j];
        c[(size_t)i * m + j] = s;
      }
  };
  auto rescale = [&](std::vector<double>& a, int& e) {
    if (a[(size_t)(k - 1) * m + (k - 1)] > 1e140) {
      for (auto& v : a) v *= 1e-140;
      e += 140;
    }
  };
  // initialize Q = I
  for (int i = 0; i < m; ++i) Q[(size_t)i * m + i] = 1.0;
  int nn = n;
  while (nn > 0) {
    if (nn & 1) {
      matmul(Q, H, tmp);
      Q.swap(tmp);
      eQ += eH;
      rescale(Q, eQ);
    }
    matmul(H, H, tmp);
    H.swap(tmp);
    eH *= 2;
    rescale(H, eH);
    nn >>= 1;
  }
  double s = Q[(size_t)(k - 1) * m + (k - 1)];
  // multiply by n! / n^n with the same scaling discipline
  for (int i = 1; i <= n; ++i) {
    s *= (double)i / n;
    if (s < 1e-140) {
      s *= 1e140;
      eQ -= 140;
    }
  }
  const double cdf = s * std::pow(10.0, eQ);
  return std::min(1.0, std::max(0.0, 1.0 - cdf));
}

static double pelz_good_sf(double x, double n) {
  if (x <= 0.0) return 1.0;
  if (x >= 1.0) return 0.0;
  const double z = std::sqrt(n) * x;
  const double z2 = z * z, z3 = z2 * z, z4 = z2 * z2, z6 = z4 * z2;
  const double z7 = z6 * z, z8 = z4 * z4, z10 = z8 * z2;
  const double PI2 = M_PI * M_PI, PI4 = PI2 * PI2, PI6 = PI4 * PI2;
  const double SQRT2PI = std::sqrt(2.0 * M_PI);

  const double qlog = -PI2 / 8.0 / z2;
  if (qlog < -690.0) return 1.0;  // cdf ~ 0
  const double q = std::exp(qlog);

  const double k1a = -z2, k1b = PI2 / 4.0;
  const double k2a = 6 * z6 + 2 * z4;
  const double k2b = (2 * z4 - 5 * z2) * PI2 / 4.0;
  const double k2c = PI4 * (1 - 2 * z2) / 16.0;
  const double k3d = PI6 * (5 - 30 * z2) / 64.0;
  const double k3c = PI4 * (-60 * z2 + 212 * z4) / 16.0;
  const double k3b = PI2 * (135 * z4 - 96 * z6) / 4.0;
  const double k3a = -30 * z6 - 90 * z8;

  double K0 = 0, K1 = 0, K2 = 0, K3 = 0;
  const int maxk = (int)std::ceil(16.0 * z / M_PI);
  for (int k = maxk; k >= 1; --k) {
    const double m = 2.0 * k - 1.0;
    const double m2 = m * m, m4 = m2 * m2, m6 = m4 * m2;
    const double qp = std::pow(q, 8.0 * k);
    K0 = K0 * qp + 1.0;
    K1 = K1 * qp + (k1a + k1b * m2);
    K2 = K2 * qp + (k2a + k2b * m2 + k2c * m4);
    K3 = K3 * qp + (k3a + k3b * m2 + k3c * m4 + k3d * m6);
  }
  K0 *= q * SQRT2PI / z;
  K1 *= q * SQRT2PI / (6 * z4);
  K2 *= q * SQRT2PI / (72 * z7);
  K3 *= q * SQRT2PI / (6480 * z10);

  const double q2 = std::exp(-PI2 / 2.0 / z2);
  double k2x = 0, k3x = 0;
  for (int k = 1; k <= maxk; ++k) {
    const double k2_ = (double)k * k;
    const double qpw = std::pow(q2, k2_);
    k2x += k2_ * qpw;
    const double kspi = M_PI * k;
    const double s3z = std::sqrt(3.0) * z;
    k3x += (s3z + kspi) * (s3z - kspi) * k2_ * qpw;
  }
  K2 += k2x * PI2 * SQRT2PI / (-36 * z3);
  K3 += k3x * PI2 * SQRT2PI / (216 * z6);

  const double sq = std::sqrt(n);
  const double cdf = K0 + K1 / sq + K2 / n + K3 / (n * sq);
  return std::min(1.0, std::max(0.0, 1.0 - cdf));
}

static void drift_pvals_raw(const int32_t* bh, const float* kd, int nnum,
                            const int32_t* rc, const int32_t* off, int ncat,
                            int64_t n_ref, int64_t n_batch, double* pv) {

  for (int j = 0; j < ncat; ++j) {
    double rsum = 0, bsum = 0;
    int k = 0;
    const int lo = off[j], hi = off[j + 1];
    for (int i = lo; i < hi; ++i) {
      if (rc[i] + bh[i] > 0) { ++k; rsum += rc[i]; bsum += bh[i]; }
    }
    if (k < 2 || rsum == 0 || bsum == 0) { pv[j] = 1.0; continue; }
    const double n = rsum + bsum;
    double stat = 0;
    for (int i = lo; i < hi; ++i) {
      const double tot = (double)rc[i] + bh[i];
      if (tot <= 0) continue;
      const double er = tot * (rsum / n), eb = tot * (bsum / n);
      double dr = std::fabs(rc[i] - er), db = std::fabs(bh[i] - eb);
      if (k == 2) {  // Yates continuity correction on 2x2
        dr = std::max(dr - 0.5, 0.0);
        db = std::max(db - 0.5, 0.0);
      }
      stat += dr * dr / er + db * db / eb;
    }
    pv[j] = chi2_sf_int_dof(stat, k - 1);
  }

  const double en_f = (double)n_ref * (double)n_batch / ((double)n_ref + (double)n_batch);
  const double en = std::nearbyint(en_f);
  // en <= 140: exact MTW (matches scipy's exact small-n path);
  // en > 140: Pelz-Good series (<= 3e-7 abs vs exact, tested)
  if (en <= 140.0) {
    for (int j = 0; j < nnum; ++j) pv[ncat + j] = mtw_sf((int)en, (double)kd[j]);
  } else {
    for (int j = 0; j < nnum; ++j) pv[ncat + j] = pelz_good_sf((double)kd[j], en);
  }
}

py::array_t<double> drift_pvals_host(
    py::array_t<int32_t> batch_hist, py::array_t<float> ks_d,
    py::array_t<int32_t> ref_cat_counts, py::array_t<int32_t> cat_offsets,
    int64_t n_ref, int64_t n_batch) {
  const int ncat = (int)cat_offsets.size() - 1;
  const int nnum = (int)ks_d.size();
  py::array_t<double> out({(ssize_t)(ncat + nnum)});
  drift_pvals_raw(batch_hist.data(), ks_d.data(), nnum, ref_cat_counts.data(),
                  cat_offsets.data(), ncat, n_ref, n_batch, out.mutable_data());
  return out;
}

py::tuple encode_json(py::bytes body, py::list vocabs, py::list cat_names,
                      py::list num_names, py::str missing_cat,
                      py::array_t<int16_t> default_codes,
                      py::array_t<float> default_nums) {
  JsonEncoderState st(vocabs, cat_names, num_names, missing_cat,
                      default_codes, default_nums);
  return encode_json_impl(st, body);
}

// ---------------------------------------------------------------------------
// Response serializer: the reference-shaped /score response straight to
// JSON bytes (shortest-round-trip doubles via std::to_chars). Skips the
// Python dict + json.dumps pass on the serving hot path.
// ---------------------------------------------------------------------------

static inline void append_double(std::string& out, double v) {
  char buf[32];
  auto [p, ec] = std::to_chars(buf, buf + sizeof(buf), v);
  (void)ec;
  // JSON requires a fraction or exponent for floats; Python emits "0.0"
  // style. Keep parity: append ".0" to bare integers.
  bool has_dot = false;
  for (char* c = buf; c < p; ++c)
    if (*c == '.' || *c == 'e' || *c == 'E' || *c == 'n' || *c == 'i') { has_dot = true; break; }
  out.append(buf, p);
  if (!has_dot) out += ".0";
}

py::bytes build_response_json(py::object pin_outs, int64_t b,
                              py::array_t<double> pvals, py::list feature_names) {
  auto outs = py::cast<torch::Tensor>(pin_outs);
  const double* base = outs.data_ptr<double>();
  const double* proba = base;
  const double* outlier = base + 2 * b;
  const int nf = (int)py::len(feature_names);
  const double* pv = pvals.data();

  std::vector<std::string> names(nf);
  for (int j = 0; j < nf; ++j) names[j] = py::cast<std::string>(feature_names[j]);

  std::string out;
  {
    py::gil_scoped_release nogil;
    out.reserve((size_t)b * 24 + 2048);
    out += "{\"predictions\": [";
    for (int64_t i = 0; i < b; ++i) {
      if (i) out += ", ";
      append_double(out, proba[i]);
    }
    out += "], \"outliers\": [";
    for (int64_t i = 0; i < b; ++i) {
      if (i) out += ", ";
      out += (outlier[i] != 0.0) ? "1.0" : "0.0";
    }
    out += "], \"feature_drift_batch\": {";
    for (int j = 0; j < nf; ++j) {
      if (j) out += ", ";
      out += '\"';
      out += names[j];
      out += "\": ";
      // (1 - p) computed in float32, reference 02-register cell-9 semantics
      append_double(out, (double)(1.0f - (float)pv[j]));
    }
    out += "}}";
  }
  return py::bytes(out);
}

// Serialize a /score response from plain arrays (the micro-batcher's
// merged-flush slices) — same wire bytes as build_response_json but not
// tied to the session's pinned layout. GIL released during the build.
py::bytes build_response_json_arrays(py::array_t<double> predictions,
                                     py::array_t<double> outliers,
                                     py::array_t<double> pvals,
                                     py::list feature_names) {
  const double* proba = predictions.data();
  const double* outl = outliers.data();
  const double* pv = pvals.data();
  const int64_t b = (int64_t)predictions.size();
  TORCH_CHECK((int64_t)outliers.size() == b, "outliers size mismatch");
  const int nf = (int)py::len(feature_names);
  TORCH_CHECK((int)pvals.size() == nf, "pvals size mismatch");
  std::vector<std::string> names(nf);
  for (int j = 0; j < nf; ++j) names[j] = py::cast<std::string>(feature_names[j]);

  std::string out;
  {
    py::gil_scoped_release nogil;
    out.reserve((size_t)b * 24 + 2048);
    out += "{\"predictions\": [";
    for (int64_t i = 0; i < b; ++i) {
      if (i) out += ", ";
      append_double(out, proba[i]);
    }
    out += "], \"outliers\": [";
    for (int64_t i = 0; i < b; ++i) {
      if (i) out += ", ";
      out += (outl[i] != 0.0) ? "1.0" : "0.0";
    }
    out += "], \"feature_drift_batch\": {";
    for (int j = 0; j < nf; ++j) {
      if (j) out += ", ";
      out += '\"';
      out += names[j];
      out += "\": ";
      append_double(out, (double)(1.0f - (float)pv[j]));
    }
    out += "}}";
  }
  return py::bytes(out);
}

// ---------------------------------------------------------------------------
// The consolidated request path: one call = parse wire JSON -> pinned
// staging -> graph replay -> drift p-values -> response JSON bytes. No
// Python object pass at any stage (GIL released around parse/serialize;
// ScoreSession::score releases it while waiting on the GPU).
// ---------------------------------------------------------------------------

py::tuple score_json_full(ScoreSession& s, py::bytes body,
                          const JsonEncoderState& st,
                          py::array_t<int32_t> ref_cat_counts,
                          py::array_t<int32_t> cat_offsets, int64_t n_ref,
                          py::list feature_names, int64_t drift_cap) {
  char* data;
  ssize_t blen;
  if (PyBytes_AsStringAndSize(body.ptr(), &data, &blen) != 0)
    throw py::value_error("body must be bytes");
  const int nf = (int)py::len(feature_names);
  TORCH_CHECK(nf == N_CAT + N_NUM, "feature name count mismatch");
  std::vector<std::string> names(nf);
  for (int j = 0; j < nf; ++j) names[j] = py::cast<std::string>(feature_names[j]);

  std::vector<int16_t> codes;
  std::vector<float> nums;
  size_t b;
  {
    py::gil_scoped_release nogil;
    try {
      b = parse_records_nogil(st, data, blen, codes, nums);
    } catch (const jsonenc::Error& e) {
      py::gil_scoped_acquire gil;
      throw py::value_error(e.msg);
    }
  }
  if (b == 0) throw py::value_error("empty request batch");
  TORCH_CHECK((int64_t)b <= s.capacity, "batch exceeds session capacity: ", b);

  std::memcpy(s.p_codes(0), codes.data(), b * N_CAT * sizeof(int16_t));
  std::memcpy(s.p_nums(0), nums.data(), b * N_NUM * sizeof(float));
  const int64_t cap = (drift_cap > 0) ? std::min<int64_t>(drift_cap, MAX_DRIFT_ROWS)
                                      : MAX_DRIFT_ROWS;
  int64_t nb = (int64_t)b;
  if ((int64_t)b <= cap) {
    s.score((int64_t)b, true, true, 0);
  } else {
    // Oversized batch: drift is a batch-population statistic, so run the
    // capped-sample drift pass FIRST (only with_drift passes write the
    // pinned drift region), then the full batch without drift — pin_outs
    // then holds the b-packed layout the serializer below reads. Running
    // the passes the other way round overwrote the b-packed outputs with a
    // cap-packed layout and corrupted rows >= cap (round-1 advisor
    // finding).
    s.score(cap, true, true, 0);
    s.score((int64_t)b, false, true, 0);
    nb = cap;
  }

  double pv[N_CAT + N_NUM];
  drift_pvals_raw(s.p_hist(0), s.p_ksd(0), N_NUM,
                  ref_cat_counts.data(), cat_offsets.data(),
                  (int)cat_offsets.size() - 1, n_ref, nb, pv);

  std::string out;
  {
    py::gil_scoped_release nogil;
    const double* proba = s.p_outs(0);
    const double* outlier = proba + 2 * b;
    out.reserve(b * 24 + 2048);
    out += "{\"predictions\": [";
    for (size_t i = 0; i < b; ++i) {
      if (i) out += ", ";
      append_double(out, proba[i]);
    }
    out += "], \"outliers\": [";
    for (size_t i = 0; i < b; ++i) {
      if (i) out += ", ";
      out += (outlier[i] != 0.0) ? "1.0" : "0.0";
    }
    out += "], \"feature_drift_batch\": {";
    for (int j = 0; j < nf; ++j) {
      if (j) out += ", ";
      out += '\"';
      out += names[j];
      out += "\": ";
      append_double(out, (double)(1.0f - (float)pv[j]));
    }
    out += "}}";
  }
  return py::make_tuple(py::bytes(out), (int64_t)b);
}

// Stage encoded arrays into a pinned slot and launch — one C call with
// the GIL released around the memcpys (the numpy slice-assign staging it
// replaces held the GIL for ~10 µs per step).
void submit_arrays(ScoreSession& s, py::array_t<int16_t> codes,
                   py::array_t<float> nums, int64_t slot, bool with_drift,
                   bool sync) {
  TORCH_CHECK(codes.ndim() == 2 && codes.shape(1) == N_CAT, "codes must be [B, N_CAT]");
  TORCH_CHECK(nums.ndim() == 2 && nums.shape(1) == N_NUM, "nums must be [B, N_NUM]");
  const int64_t b = codes.shape(0);
  TORCH_CHECK(nums.shape(0) == b, "codes/nums row mismatch");
  TORCH_CHECK(b >= 1 && b <= s.capacity, "batch out of range: ", b);
  auto c = codes.unchecked<2>();
  auto n = nums.unchecked<2>();
  const int sl = (int)(slot & 1);
  {
    py::gil_scoped_release nogil;
    // pybind returns row-contiguous for c_style-convertible inputs; copy
    // row-wise to stay correct for any stride
    int16_t* pc = s.p_codes(sl);
    float* pn = s.p_nums(sl);
    if (codes.strides(0) == (ssize_t)(N_CAT * sizeof(int16_t)) &&
        codes.strides(1) == (ssize_t)sizeof(int16_t)) {
      std::memcpy(pc, c.data(0, 0), (size_t)b * N_CAT * sizeof(int16_t));
    } else {
      for (int64_t i = 0; i < b; ++i)
        for (int j = 0; j < N_CAT; ++j) pc[i * N_CAT + j] = c(i, j);
    }
    if (nums.strides(0) == (ssize_t)(N_NUM * sizeof(float)) &&
        nums.strides(1) == (ssize_t)sizeof(float)) {
      std::memcpy(pn, n.data(0, 0), (size_t)b * N_NUM * sizeof(float));
    } else {
      for (int64_t i = 0; i < b; ++i)
        for (int j = 0; j < N_NUM; ++j) pn[i * N_NUM + j] = n(i, j);
    }
  }
  s.score(b, with_drift, sync, sl);
}

// Pipelined epilogue: wait for slot's graph, convert drift p-values and
// serialize the response from that slot's pinned buffers.
py::bytes response_epilogue(ScoreSession& s, int64_t slot64, int64_t b64,
                            py::array_t<int32_t> ref_cat_counts,
                            py::array_t<int32_t> cat_offsets, int64_t n_ref,
                            py::list feature_names) {
  const int slot = (int)slot64 & 1;
  const int64_t b = b64;
  TORCH_CHECK(b >= 1 && b <= s.capacity, "batch out of range: ", b);
  const int nf = (int)py::len(feature_names);
  std::vector<std::string> names(nf);
  for (int j = 0; j < nf; ++j) names[j] = py::cast<std::string>(feature_names[j]);
  const int64_t nb = std::min<int64_t>(b, MAX_DRIFT_ROWS);

  double pv[N_CAT + N_NUM];
  std::string out;
  {
    py::gil_scoped_release nogil;
    sync_event(s.ev_done[slot]);
    drift_pvals_raw(s.p_hist(slot), s.p_ksd(slot), N_NUM,
                    ref_cat_counts.data(), cat_offsets.data(),
                    (int)cat_offsets.size() - 1, n_ref, nb, pv);
    const double* proba = s.p_outs(slot);
    const double* outlier = proba + 2 * b;
    out.reserve((size_t)b * 24 + 2048);
    out += "{\"predictions\": [";
    for (int64_t i = 0; i < b; ++i) {
      if (i) out += ", ";
      append_double(out, proba[i]);
    }
    out += "], \"outliers\": [";
    for (int64_t i = 0; i < b; ++i) {
      if (i) out += ", ";
      out += (outlier[i] != 0.0) ? "1.0" : "0.0";
    }
    out += "], \"feature_drift_batch\": {";
    for (int j = 0; j < nf; ++j) {
      if (j) out += ", ";
      out += '\"';
      out += names[j];
      out += "\": ";
      append_double(out, (double)(1.0f - (float)pv[j]));
    }
    out += "}}";
  }
  return py::bytes(out);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("score_forest_pipeline", &score_forest_pipeline,
        "Forest classifier + isolation forest scoring (gfx950)");
  m.def("drift_stats", &drift_stats,
        "Per-feature drift statistics: categorical histograms + K-S D (gfx950)");
  m.def("encode_records", &encode_records,
        "Native request encoder: list[dict] -> (codes i16[B,9], nums f32[B,14])");
  m.def("encode_json", &encode_json,
        "Parse a /score JSON request body straight into (codes, nums)");
  py::class_<JsonEncoderState>(m, "JsonEncoder")
      .def(py::init<py::list, py::list, py::list, py::str,
                    py::array_t<int16_t>, py::array_t<float>>())
      .def("encode", [](const JsonEncoderState& st, py::bytes body) {
        return encode_json_impl(st, body);
      });
  m.def("dense_score", &dense_score,
        "Fused impute + logistic linear score + robust-z outlier (gfx950)");
  m.def("impute_transpose", &impute_transpose,
        "Fused NaN->median impute + transpose [B,F] -> [F,B] (gfx950)");
  m.def("ks_stats_sorted", &ks_stats_sorted,
        "Exact K-S D per column for pre-sorted batch columns (gfx950)");
  m.def("forest_ilp_bench", &forest_ilp_bench,
        "A/B: forest traversal, 1 vs 2 trees per thread");
  m.def("ks_stats", &ks_stats,
        "Exact two-sample K-S D per column, any column count (gfx950)",
        py::arg("nums"), py::arg("medians"), py::arg("ref_sorted"),
        py::arg("rs_off"), py::arg("block") = 256, py::arg("ref_lds") = 0);
  m.def("build_response_json", &build_response_json,
        "Serialize the /score response to JSON bytes (C, shortest doubles)");
  m.def("build_response_json_arrays", &build_response_json_arrays,
        "Serialize a /score response from plain arrays (merged-flush slices)");
  m.def("drift_pvals_host", &drift_pvals_host,
        "Drift p-values from kernel statistics (chi2 + Pelz-Good K-S), host C");
  py::class_<ScoreSession>(m, "ScoreSession")
      .def(py::init<py::dict, int64_t, int>(), py::arg("model"),
           py::arg("capacity"), py::arg("device_index"))
      .def("score", &ScoreSession::score, py::arg("b"),
           py::arg("with_drift") = true, py::arg("sync") = true,
           py::arg("slot") = 0)
      .def("wait_slot", &ScoreSession::wait_slot)
      .def("submit_arrays", &submit_arrays, py::arg("codes"), py::arg("nums"),
           py::arg("slot") = 0, py::arg("with_drift") = true,
           py::arg("sync") = false)
      .def("response_epilogue", &response_epilogue, py::arg("slot"),
           py::arg("b"), py::arg("ref_cat_counts"), py::arg("cat_offsets"),
           py::arg("n_ref"), py::arg("feature_names"))
      .def("synchronize", &ScoreSession::synchronize)
      .def("score_json_full", &score_json_full, py::arg("body"),
           py::arg("encoder"), py::arg("ref_cat_counts"),
           py::arg("cat_offsets"), py::arg("n_ref"), py::arg("feature_names"),
           py::arg("drift_cap") = -1)
      .def_readonly("capacity", &ScoreSession::capacity)
      .def_property_readonly("pin_codes", [](ScoreSession& s) { return s.pin_codes; })
      .def_property_readonly("pin_nums", [](ScoreSession& s) { return s.pin_nums; })
      .def_property_readonly("pin_outs", [](ScoreSession& s) { return s.pin_outs; })
      .def_property_readonly("pin_hist", [](ScoreSession& s) { return s.pin_hist; })
      .def_property_readonly("pin_ksd", [](ScoreSession& s) { return s.pin_ksd; });
}
