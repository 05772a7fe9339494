// MI355X (gfx950 / CDNA4) kernels for the histogram tree updater.
//
// Replaces the CUDA gpu_hist kernel set the reference reaches through
// xgb.train: per-node gradient histograms, row partition,
// leaf scatter and batched forest prediction. Written directly for CDNA4:
// 64-wide wavefronts, 160 KiB LDS per CU, device-scope atomics, grids capped
// and grid-strided per Guideline 11 of the CDNA HIP programming guide.
//
// Histogram accumulation uses int64 fixed point (value * 2^33 / max_abs)
// in LDS with one global flush per block: bit-deterministic regardless of
// atomic ordering (the `deterministic_histogram` contract) and exactly
// summable across ranks by an RCCL int64 allreduce.
//
// All launchers are batched: one launch covers every node of a tree level,
// with a host-built block -> (job, chunk) map so the grid stays near the
// 256-CU sweet spot regardless of how many nodes/rows each level has.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <vector>

#define WAVE 64
#define HIST_BLOCK 256

// LDS histogram layout: separate g and h arrays (SoA). A 16-B interleaved
// (g,h) slot can only start on 8 of the 32 bank groups (measured 75% of
// LDS cycles lost to conflicts: SQ_LDS_BANK_CONFLICT 0.74G vs
// SQ_LDS_IDX_ACTIVE 1.0G); 8-B-strided u64 arrays reach 16 groups, halving
// intrinsic conflict pressure. lds_pad_slot adds a skew word per 8 slots.
__device__ __host__ inline int lds_pad_slot(int s) { return s + (s >> 3); }
__device__ __host__ inline long long lds_padded_words(long long pairs) {
  // per array: pairs + pairs/8 + 1 u64 words; x2 for the g and h arrays
  return (pairs + (pairs >> 3) + 1) * 2;
}
__device__ inline long long lds_half_words(long long pairs) {
  return pairs + (pairs >> 3) + 1;
}
#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be a ROCm device tensor")

// ---------------------------------------------------------------------------
// job descriptors (mirrored by host-side packing in ops/hip.py)
// ---------------------------------------------------------------------------

struct HistJob {
  int start;        // row segment [start, end) in rowbuf
  int end;
  int hist_idx;     // output histogram index
  int fg_start;     // feature group [fg_start, fg_end)
  int fg_end;
  int first_block;  // first grid block assigned to this job
  int num_blocks;
};

struct PartJob {
  int start;
  int end;
  int feature;
  int split_bin;
  int default_left;
  int first_block;
  int num_blocks;
};

struct LeafJob {
  int start;
  int end;
  int parity;
  float value;
  int first_block;
  int num_blocks;
};

// ---------------------------------------------------------------------------
// histogram build
// ---------------------------------------------------------------------------

template <typename BinT>
__global__ __launch_bounds__(HIST_BLOCK) void hist_kernel(
    const BinT* __restrict__ bins, const float2* __restrict__ gh,
    const int* __restrict__ rowbuf, const HistJob* __restrict__ jobs,
    const int* __restrict__ block_job, unsigned long long* __restrict__ out,
    int nfeat, int stride, float scale_g, float scale_h) {
  extern __shared__ unsigned long long lhist[];

  const HistJob job = jobs[block_job[blockIdx.x]];
  const int nf_group = job.fg_end - job.fg_start;
  const int lds_words = nf_group * stride * 2;
  const int hofs = (int)lds_half_words(nf_group * stride);
  const int lds_padded = 2 * hofs;
  for (int i = threadIdx.x; i < lds_padded; i += blockDim.x) lhist[i] = 0ull;
  __syncthreads();

  const int chunk = blockIdx.x - job.first_block;
  const long long step = (long long)job.num_blocks * blockDim.x;
  for (long long r = job.start + (long long)chunk * blockDim.x + threadIdx.x; r < job.end; r += step) {
    const int row = rowbuf[r];
    const float2 gp = gh[row];
    const unsigned long long gfix = (unsigned long long)(long long)llrintf(gp.x * scale_g);
    const unsigned long long hfix = (unsigned long long)(long long)llrintf(gp.y * scale_h);
    const BinT* rp = bins + (long long)row * nfeat + job.fg_start;
    if constexpr (sizeof(BinT) == 1) {
      // vectorized path: 4 bins per dword load (valid when the group is
      // 4-aligned in the row — guaranteed by host packing for nfeat%4==0)
      if ((nf_group & 3) == 0 && ((((long long)row * nfeat + job.fg_start) & 3) == 0)) {
        const uchar4* rp4 = reinterpret_cast<const uchar4*>(rp);
        #pragma unroll 2
        for (int f4 = 0; f4 < (nf_group >> 2); ++f4) {
          const uchar4 b4 = rp4[f4];
          const int base = (f4 << 2) * stride;
          const int s0 = lds_pad_slot(base + (int)b4.x);
          const int s1 = lds_pad_slot(base + stride + (int)b4.y);
          const int s2 = lds_pad_slot(base + 2 * stride + (int)b4.z);
          const int s3 = lds_pad_slot(base + 3 * stride + (int)b4.w);
          atomicAdd(&lhist[s0], gfix);
          atomicAdd(&lhist[hofs + s0], hfix);
          atomicAdd(&lhist[s1], gfix);
          atomicAdd(&lhist[hofs + s1], hfix);
          atomicAdd(&lhist[s2], gfix);
          atomicAdd(&lhist[hofs + s2], hfix);
          atomicAdd(&lhist[s3], gfix);
          atomicAdd(&lhist[hofs + s3], hfix);
        }
        continue;
      }
    }
    #pragma unroll 4
    for (int f = 0; f < nf_group; ++f) {
      const int slot = lds_pad_slot(f * stride + (int)rp[f]);
      atomicAdd(&lhist[slot], gfix);
      atomicAdd(&lhist[hofs + slot], hfix);
    }
  }
  __syncthreads();

  unsigned long long* gout =
      out + ((long long)job.hist_idx * nfeat + job.fg_start) * (long long)stride * 2;
  for (int i = threadIdx.x; i < lds_words; i += blockDim.x) {
    const int pair = i >> 1;
    const int idx = lds_pad_slot(pair) + ((i & 1) ? hofs : 0);
    const unsigned long long v = lhist[idx];
    if (v) atomicAdd(&gout[i], v);
  }
}

__global__ void hist_convert_kernel(const unsigned long long* __restrict__ in,
                                    float* __restrict__ out, long long n_pairs,
                                    float inv_g, float inv_h) {
  const long long step = (long long)gridDim.x * blockDim.x;
  const long long i0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  for (long long i = i0; i < n_pairs; i += step) {
    const long long j = i * 2;
    out[j] = (float)((double)(long long)in[j] * (double)inv_g);
    out[j + 1] = (float)((double)(long long)in[j + 1] * (double)inv_h);
  }
}

__global__ void hist_convert_dev_kernel(const unsigned long long* __restrict__ in,
                                        float* __restrict__ out, long long n_pairs,
                                        const float* __restrict__ gh_max) {
  const double inv_g = (double)fmaxf(gh_max[0], 1e-30f) / 8589934592.0;
  const double inv_h = (double)fmaxf(gh_max[1], 1e-30f) / 8589934592.0;
  const long long step = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n_pairs; i += step) {
    const long long j = i * 2;
    out[j] = (float)((double)(long long)in[j] * (double)inv_g);
    out[j + 1] = (float)((double)(long long)in[j + 1] * (double)inv_h);
  }
}

// ---------------------------------------------------------------------------
// COMPACT-LAYOUT pipeline (v2)
//
// The gather-based hist kernel reads each scattered row through a full
// 128-B cache line (28 useful bytes): at depth>0 the updater is bound by
// that amplification. v2 keeps rows PHYSICALLY node-contiguous: partition
// rewrites each surviving row's (row id, bin bytes, gradient pair) into
// ping-pong compact SoA buffers, so every level's histogram build is a
// perfectly coalesced stream and no kernel gathers by row id again.
// ---------------------------------------------------------------------------

// hist over compact buffers: rows are positions [start,end) in bins_c/gh_c.
template <typename BinT>
__global__ __launch_bounds__(HIST_BLOCK) void hist_compact_kernel(
    const BinT* __restrict__ bins_c, const float2* __restrict__ gh_c,
    const HistJob* __restrict__ jobs, const int* __restrict__ block_job,
    unsigned long long* __restrict__ out, int nfeat, int stride,
    const float* __restrict__ gh_max) {
  extern __shared__ unsigned long long lhist[];

  // fixed-point scale derived on device (2^33 / max_abs): no host sync
  const float scale_g = 8589934592.0f / fmaxf(gh_max[0], 1e-30f);
  const float scale_h = 8589934592.0f / fmaxf(gh_max[1], 1e-30f);
  const HistJob job = jobs[block_job[blockIdx.x]];
  const int nf_group = job.fg_end - job.fg_start;
  const int lds_words = nf_group * stride * 2;
  const int hofs = (int)lds_half_words(nf_group * stride);
  const int lds_padded = 2 * hofs;
  for (int i = threadIdx.x; i < lds_padded; i += blockDim.x) lhist[i] = 0ull;
  __syncthreads();

  const int chunk = blockIdx.x - job.first_block;
  const long long step = (long long)job.num_blocks * blockDim.x;
  for (long long r = job.start + (long long)chunk * blockDim.x + threadIdx.x; r < job.end; r += step) {
    const float2 gp = gh_c[r];
    const unsigned long long gfix = (unsigned long long)(long long)llrintf(gp.x * scale_g);
    const unsigned long long hfix = (unsigned long long)(long long)llrintf(gp.y * scale_h);
    const BinT* rp = bins_c + (long long)r * nfeat + job.fg_start;
    if constexpr (sizeof(BinT) == 1) {
      if ((nf_group & 3) == 0 && ((((long long)r * nfeat + job.fg_start) & 3) == 0)) {
        const uchar4* rp4 = reinterpret_cast<const uchar4*>(rp);
        #pragma unroll 2
        for (int f4 = 0; f4 < (nf_group >> 2); ++f4) {
          const uchar4 b4 = rp4[f4];
          const int base = (f4 << 2) * stride;
          const int s0 = lds_pad_slot(base + (int)b4.x);
          const int s1 = lds_pad_slot(base + stride + (int)b4.y);
          const int s2 = lds_pad_slot(base + 2 * stride + (int)b4.z);
          const int s3 = lds_pad_slot(base + 3 * stride + (int)b4.w);
          atomicAdd(&lhist[s0], gfix);
          atomicAdd(&lhist[hofs + s0], hfix);
          atomicAdd(&lhist[s1], gfix);
          atomicAdd(&lhist[hofs + s1], hfix);
          atomicAdd(&lhist[s2], gfix);
          atomicAdd(&lhist[hofs + s2], hfix);
          atomicAdd(&lhist[s3], gfix);
          atomicAdd(&lhist[hofs + s3], hfix);
        }
        continue;
      }
    }
    #pragma unroll 4
    for (int f = 0; f < nf_group; ++f) {
      const int slot = lds_pad_slot(f * stride + (int)rp[f]);
      atomicAdd(&lhist[slot], gfix);
      atomicAdd(&lhist[hofs + slot], hfix);
    }
  }
  __syncthreads();

  unsigned long long* gout =
      out + ((long long)job.hist_idx * nfeat + job.fg_start) * (long long)stride * 2;
  for (int i = threadIdx.x; i < lds_words; i += blockDim.x) {
    const int pair = i >> 1;
    const int idx = lds_pad_slot(pair) + ((i & 1) ? hofs : 0);
    const unsigned long long v = lhist[idx];
    if (v) atomicAdd(&gout[i], v);
  }
}

// compact partition: decide per row from src bins (sequential read), then
// copy the whole record (row id + bin bytes + gh) to its contiguous
// destination run. Destinations inside one tile form two coalesced runs.
#define CPART_TILE 2048

// jobs carry segments only; the split decision (gain, feature, bin,
// missing direction) is read from the on-device packed split tensor
// ([k, 6] float32 from the split kernels), so no host round-trip sits
// between split search and partition. gain <= 0 jobs are skipped.
template <typename BinT>
__global__ __launch_bounds__(HIST_BLOCK) void partition_compact_kernel(
    const BinT* __restrict__ src_bins, const float2* __restrict__ src_gh,
    const int* __restrict__ src_rows, BinT* __restrict__ dst_bins,
    float2* __restrict__ dst_gh, int* __restrict__ dst_rows,
    const PartJob* __restrict__ jobs, const int* __restrict__ block_job,
    const float* __restrict__ split_packed,  // [n_nodes, 6]; job.feature = node row
    int* __restrict__ counters, int nfeat, int missing_bin) {
  __shared__ int ldest[CPART_TILE];  // destination index per tile row
  __shared__ int lcnt, rcnt, lbase, rbase;

  const int j = block_job[blockIdx.x];
  const PartJob job = jobs[j];
  const float* sp6 = split_packed + job.feature * 6;
  if (sp6[0] <= 0.0f) return;  // no split for this node
  const int feature = (int)sp6[1];
  const int split_bin = (int)sp6[2];
  const int default_left = sp6[3] > 0.5f ? 1 : 0;

  const int chunk = blockIdx.x - job.first_block;
  const long long tile_step = (long long)job.num_blocks * CPART_TILE;

  for (long long tile = job.start + (long long)chunk * CPART_TILE; tile < job.end; tile += tile_step) {
    if (threadIdx.x == 0) {
      lcnt = 0;
      rcnt = 0;
    }
    __syncthreads();
    const int tile_n = (int)min((long long)CPART_TILE, job.end - tile);
    for (int i = threadIdx.x; i < tile_n; i += blockDim.x) {
      const int b = (int)src_bins[(tile + i) * (long long)nfeat + feature];
      const bool left = (b == missing_bin) ? (default_left != 0) : (b <= split_bin);
      ldest[i] = left ? atomicAdd(&lcnt, 1) : ~atomicAdd(&rcnt, 1);
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      lbase = atomicAdd(&counters[j * 2], lcnt);
      rbase = atomicAdd(&counters[j * 2 + 1], rcnt);
    }
    __syncthreads();
    // copy phase, (row, dword) work units: consecutive threads move
    // consecutive dwords so stores coalesce into the two destination runs.
    // Work units are padded to a power of two per row so the row index is a
    // shift, not an integer division (a div per dword dominated this loop).
    if ((nfeat & 3) == 0 && sizeof(BinT) == 1) {
      const int nd = nfeat >> 2;  // bin dwords per row
      int log2p = 0;
      while ((1 << log2p) < nd) ++log2p;
      const int mask = (1 << log2p) - 1;
      const uchar4* sb4 = reinterpret_cast<const uchar4*>(src_bins);
      uchar4* db4 = reinterpret_cast<uchar4*>(dst_bins);
      for (int u = threadIdx.x; u < (tile_n << log2p); u += blockDim.x) {
        const int i = u >> log2p;
        const int f4 = u & mask;
        if (f4 >= nd) continue;
        const int d = ldest[i];
        const long long dst =
            d >= 0 ? (long long)job.start + lbase + d : (long long)job.end - 1 - rbase - (~d);
        db4[dst * nd + f4] = sb4[(tile + i) * (long long)nd + f4];
      }
    } else {
      int log2p = 0;
      while ((1 << log2p) < nfeat) ++log2p;
      const int mask = (1 << log2p) - 1;
      for (int u = threadIdx.x; u < (tile_n << log2p); u += blockDim.x) {
        const int i = u >> log2p;
        const int f = u & mask;
        if (f >= nfeat) continue;
        const int d = ldest[i];
        const long long dst =
            d >= 0 ? (long long)job.start + lbase + d : (long long)job.end - 1 - rbase - (~d);
        dst_bins[dst * (long long)nfeat + f] = src_bins[(tile + i) * (long long)nfeat + f];
      }
    }
    for (int u = threadIdx.x; u < tile_n * 2; u += blockDim.x) {
      const int i = u >> 1;
      const int half = u & 1;
      const int d = ldest[i];
      const long long dst =
          d >= 0 ? (long long)job.start + lbase + d : (long long)job.end - 1 - rbase - (~d);
      reinterpret_cast<float*>(dst_gh)[dst * 2 + half] =
          reinterpret_cast<const float*>(src_gh)[(tile + i) * 2 + half];
    }
    for (int i = threadIdx.x; i < tile_n; i += blockDim.x) {
      const int d = ldest[i];
      const long long dst =
          d >= 0 ? (long long)job.start + lbase + d : (long long)job.end - 1 - rbase - (~d);
      dst_rows[dst] = src_rows[tile + i];
    }
    __syncthreads();
  }
}

// fused per-round gradient computation: one HBM pass writes the packed
// (g, h) pairs AND per-block |g|/|h| maxima (for the fixed-point histogram
// scale), replacing ~5 torch elementwise/reduce passes (~250 us/round on the
// 12.5M-row bench; this kernel is ~35 us).
// mode 0: binary:logistic / reg:logistic — g = sigmoid(m) - y,
//         h = max(p(1-p), 1e-16), scale_pos_weight applied to y == 1
// mode 1: reg:squarederror — g = m - y, h = 1
__global__ __launch_bounds__(HIST_BLOCK) void grad_fused_kernel(
    const float* __restrict__ margin, const float* __restrict__ y,
    const float* __restrict__ w, float2* __restrict__ gh,
    float2* __restrict__ pmax, double2* __restrict__ psum, long long n, int mode,
    float spw) {
  __shared__ float red[HIST_BLOCK * 2];
  __shared__ double dred[HIST_BLOCK * 2];
  float gmax = 0.f, hmax = 0.f;
  double gsum = 0.0, hsum = 0.0;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const float m = margin[i];
    const float yi = y[i];
    float g, h;
    if (mode == 0) {
      const float p = 1.0f / (1.0f + expf(-m));
      g = p - yi;
      h = fmaxf(p * (1.0f - p), 1e-16f);
      if (spw != 1.0f && yi == 1.0f) {
        g *= spw;
        h *= spw;
      }
    } else {
      g = m - yi;
      h = 1.0f;
    }
    if (w) {
      const float wi = w[i];
      g *= wi;
      h *= wi;
    }
    gh[i] = make_float2(g, h);
    gmax = fmaxf(gmax, fabsf(g));
    hmax = fmaxf(hmax, fabsf(h));
    gsum += (double)g;
    hsum += (double)h;
  }
  red[threadIdx.x] = gmax;
  red[threadIdx.x + HIST_BLOCK] = hmax;
  dred[threadIdx.x] = gsum;
  dred[threadIdx.x + HIST_BLOCK] = hsum;
  __syncthreads();
  for (int off = HIST_BLOCK / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) {
      red[threadIdx.x] = fmaxf(red[threadIdx.x], red[threadIdx.x + off]);
      red[threadIdx.x + HIST_BLOCK] =
          fmaxf(red[threadIdx.x + HIST_BLOCK], red[threadIdx.x + HIST_BLOCK + off]);
      dred[threadIdx.x] += dred[threadIdx.x + off];
      dred[threadIdx.x + HIST_BLOCK] += dred[threadIdx.x + HIST_BLOCK + off];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    pmax[blockIdx.x] = make_float2(red[0], red[HIST_BLOCK]);
    psum[blockIdx.x] = make_double2(dred[0], dred[HIST_BLOCK]);
  }
}

// leaf scatter from compact row-id buffers. The random 4-B RMW is
// latency-bound; 4 rows in flight per iteration overlap the line fetches
// (leaf row sets are disjoint by construction, so the += needs no atomics).
__global__ __launch_bounds__(HIST_BLOCK) void leaf_update_compact_kernel(
    const int* __restrict__ rows0, const int* __restrict__ rows1,
    float* __restrict__ margin, const LeafJob* __restrict__ jobs,
    const int* __restrict__ block_job, long long col_stride) {
  const LeafJob job = jobs[block_job[blockIdx.x]];
  const int* src = job.parity ? rows1 : rows0;
  const int chunk = blockIdx.x - job.first_block;
  const long long step = (long long)job.num_blocks * blockDim.x;
  long long r = job.start + (long long)chunk * blockDim.x + threadIdx.x;
  constexpr int LROWS = 4;
  for (; r + (LROWS - 1) * step < job.end; r += LROWS * step) {
    int rid[LROWS];
    #pragma unroll
    for (int u = 0; u < LROWS; ++u) rid[u] = src[r + u * step];
    #pragma unroll
    for (int u = 0; u < LROWS; ++u) margin[(long long)rid[u] * col_stride] += job.value;
  }
  for (; r < job.end; r += step) {
    margin[(long long)src[r] * col_stride] += job.value;
  }
}

// ---------------------------------------------------------------------------
// DEVICE-AUTONOMOUS depthwise grow (v3)
//
// The per-level host round-trip (drain -> python job packing -> launches)
// costs ~2-3 ms of a 7 ms round. v3 removes it: the whole tree's kernel
// sequence is enqueued up front; a tiny single-block `make_level` kernel
// builds the next level's node table and block-assignment prefixes ON
// DEVICE, hist/partition kernels walk virtual-block work lists found by
// binary search, and the host reads back ONE packed buffer per tree
// (split records + child counts + node sums) to build the tree object.
//
// Heap indexing: level d slot i lives at heap index (2^d - 1 + i);
// children of (d, i) are (d+1, 2i) and (d+1, 2i+1). Dead slots have
// start == end.
// ---------------------------------------------------------------------------

struct LevelNode {
  int start;
  int end;
  int build;  // 1 = histogram built from rows, 0 = derived by subtraction
};

// prefix tables for one level's virtual-block work distribution
struct LevelWork {
  int hist_total;   // total hist virtual blocks
  int part_total;   // total partition virtual blocks
};

#define GROW_MAX_SLOTS 1024  // max nodes per level for the device driver

// Single-block kernel: build level d+1's node table from level d's
// splits/counts, choose build children (smaller hessian), compute
// per-slot virtual-block counts + exclusive prefixes for hist & partition.
__global__ __launch_bounds__(GROW_MAX_SLOTS) void make_level_kernel(
    const LevelNode* __restrict__ cur,    // [k] level d nodes
    const float* __restrict__ splits,     // [k, 6] level d split records
    const int* __restrict__ counts,       // [k, 2] level d partition counts
    const float* __restrict__ node_gh,    // [k, 2] level d node sums
    LevelNode* __restrict__ nxt,          // [2k] level d+1 nodes
    float* __restrict__ nxt_gh,           // [2k, 2]
    int* __restrict__ hist_prefix,        // [2k + 1]
    int* __restrict__ part_prefix,        // [2k + 1]
    LevelWork* __restrict__ work, int k, int rows_per_block, int max_blocks) {
  __shared__ int h_counts[2 * GROW_MAX_SLOTS];
  __shared__ int p_counts[2 * GROW_MAX_SLOTS];
  const int tid = threadIdx.x;

  for (int i = tid; i < k; i += blockDim.x) {
    const float* sp = splits + i * 6;
    const LevelNode node = cur[i];
    LevelNode lft = {0, 0, 0}, rgt = {0, 0, 0};
    float lg = 0.f, lh = 0.f, rg = 0.f, rh = 0.f;
    if (node.end > node.start && sp[0] > 0.0f) {
      const int lc = counts[i * 2];
      lft.start = node.start;
      lft.end = node.start + lc;
      rgt.start = lft.end;
      rgt.end = node.end;
      lg = sp[4];
      lh = sp[5];
      rg = node_gh[i * 2] - lg;
      rh = node_gh[i * 2 + 1] - lh;
      // subtraction trick: build the smaller-hessian child
      if (lh <= rh) {
        lft.build = 1;
      } else {
        rgt.build = 1;
      }
    }
    nxt[2 * i] = lft;
    nxt[2 * i + 1] = rgt;
    nxt_gh[4 * i] = lg;
    nxt_gh[4 * i + 1] = lh;
    nxt_gh[4 * i + 2] = rg;
    nxt_gh[4 * i + 3] = rh;
    for (int c = 0; c < 2; ++c) {
      const LevelNode child = c ? rgt : lft;
      const int rows = child.end - child.start;
      h_counts[2 * i + c] =
          (child.build && rows > 0)
              ? (int)min((long long)(rows + rows_per_block - 1) / rows_per_block, (long long)max_blocks)
              : 0;
      p_counts[2 * i + c] =
          rows > 0 ? (int)min((long long)(rows + rows_per_block - 1) / rows_per_block,
                              (long long)max_blocks)
                   : 0;
    }
  }
  __syncthreads();
  if (tid == 0) {  // serial scan: 2k <= 2048 entries, negligible
    int hs = 0, ps = 0;
    for (int i = 0; i < 2 * k; ++i) {
      hist_prefix[i] = hs;
      part_prefix[i] = ps;
      hs += h_counts[i];
      ps += p_counts[i];
    }
    hist_prefix[2 * k] = hs;
    part_prefix[2 * k] = ps;
    work->hist_total = hs;
    work->part_total = ps;
  }
}

// Root bootstrap: one node covering [0, cap); also its hist job prefix.
__global__ void make_root_kernel(LevelNode* root, int* hist_prefix, int* part_prefix,
                                 LevelWork* work, int cap, int rows_per_block, int max_blocks) {
  root->start = 0;
  root->end = cap;
  root->build = 1;
  const int nb = (int)min((long long)(cap + rows_per_block - 1) / rows_per_block, (long long)max_blocks);
  hist_prefix[0] = 0;
  hist_prefix[1] = nb;
  part_prefix[0] = 0;
  part_prefix[1] = nb;
  work->hist_total = nb;
  work->part_total = nb;
}

__device__ inline int find_slot(const int* __restrict__ prefix, int k, int vb) {
  int lo = 0, hi = k;  // find i with prefix[i] <= vb < prefix[i+1]
  while (lo + 1 < hi) {
    const int mid = (lo + hi) >> 1;
    if (prefix[mid] <= vb) {
      lo = mid;
    } else {
      hi = mid;
    }
  }
  return lo;
}

// hist over compact buffers driven by a device job table.
// BLOCK=256 pairs with grouped 56 KB slabs (2 blocks/CU); BLOCK=512 runs a
// SINGLE full-feature slab (up to ~158 KB, 1 block/CU — the same 8 waves,
// but bins/gh stream ONCE per row instead of once per feature group).
template <typename BinT, int BLOCK>
__global__ __launch_bounds__(BLOCK) void hist_device_kernel(
    const BinT* __restrict__ bins_c, const float2* __restrict__ gh_c,
    const LevelNode* __restrict__ nodes, const int* __restrict__ hist_prefix,
    const LevelWork* __restrict__ work, unsigned long long* __restrict__ out,
    int k, int nfeat, int stride, int n_groups, int feats_per_group,
    const float* __restrict__ gh_max, int rows_per_block, int slot_lo, int slot_hi) {
  extern __shared__ unsigned long long lhist[];
  const float scale_g = 8589934592.0f / fmaxf(gh_max[0], 1e-30f);
  const float scale_h = 8589934592.0f / fmaxf(gh_max[1], 1e-30f);
  const int total = work->hist_total * n_groups;

  for (int vb = blockIdx.x; vb < total; vb += gridDim.x) {
    const int fg = vb % n_groups;
    const int hvb = vb / n_groups;
    const int slot = find_slot(hist_prefix, k, hvb);
    if (slot < slot_lo || slot >= slot_hi) continue;  // comm-overlap slot window
    const int chunk = hvb - hist_prefix[slot];
    const int nb = hist_prefix[slot + 1] - hist_prefix[slot];
    const LevelNode node = nodes[slot];
    const int fg_start = fg * feats_per_group;
    const int nf_group = min(feats_per_group, nfeat - fg_start);
    const int lds_words = nf_group * stride * 2;
    const int hofs = (int)lds_half_words(nf_group * stride);
    const int lds_padded = 2 * hofs;

    for (int i = threadIdx.x; i < lds_padded; i += blockDim.x) lhist[i] = 0ull;
    __syncthreads();

    const long long step = (long long)nb * blockDim.x;
    const bool vec4 = sizeof(BinT) == 1 && (nf_group & 3) == 0 && (nfeat & 3) == 0 &&
                      (fg_start & 3) == 0;
    long long r = node.start + (long long)chunk * blockDim.x + threadIdx.x;
    if (vec4) {
      // 4 bins per dword load (rows are 4-aligned when nfeat % 4 == 0);
      // HROWS rows in flight per iteration so the gathers overlap
      constexpr int HROWS = 4;
      const int nd = nf_group >> 2;
      for (; r + (HROWS - 1) * step < node.end; r += HROWS * step) {
        unsigned long long gf[HROWS], hf[HROWS];
        const uchar4* rp[HROWS];
        #pragma unroll
        for (int u = 0; u < HROWS; ++u) {
          const long long ru = r + u * step;
          const float2 gp = gh_c[ru];
          gf[u] = (unsigned long long)(long long)llrintf(gp.x * scale_g);
          hf[u] = (unsigned long long)(long long)llrintf(gp.y * scale_h);
          rp[u] = reinterpret_cast<const uchar4*>(bins_c + ru * nfeat + fg_start);
        }
        #pragma unroll 2
        for (int f4 = 0; f4 < nd; ++f4) {
          const int base = (f4 << 2) * stride;
          uchar4 b4[HROWS];
          #pragma unroll
          for (int u = 0; u < HROWS; ++u) b4[u] = rp[u][f4];
          #pragma unroll
          for (int u = 0; u < HROWS; ++u) {
            const int s0 = lds_pad_slot(base + (int)b4[u].x);
            const int s1 = lds_pad_slot(base + stride + (int)b4[u].y);
            const int s2 = lds_pad_slot(base + 2 * stride + (int)b4[u].z);
            const int s3 = lds_pad_slot(base + 3 * stride + (int)b4[u].w);
            atomicAdd(&lhist[s0], gf[u]);
            atomicAdd(&lhist[hofs + s0], hf[u]);
            atomicAdd(&lhist[s1], gf[u]);
            atomicAdd(&lhist[hofs + s1], hf[u]);
            atomicAdd(&lhist[s2], gf[u]);
            atomicAdd(&lhist[hofs + s2], hf[u]);
            atomicAdd(&lhist[s3], gf[u]);
            atomicAdd(&lhist[hofs + s3], hf[u]);
          }
        }
      }
    }
    for (; r < node.end; r += step) {
      const float2 gp = gh_c[r];
      const unsigned long long gfix = (unsigned long long)(long long)llrintf(gp.x * scale_g);
      const unsigned long long hfix = (unsigned long long)(long long)llrintf(gp.y * scale_h);
      const BinT* rp = bins_c + (long long)r * nfeat + fg_start;
      if (vec4) {
        const uchar4* rp4 = reinterpret_cast<const uchar4*>(rp);
        #pragma unroll 2
        for (int f4 = 0; f4 < (nf_group >> 2); ++f4) {
          const uchar4 b4 = rp4[f4];
          const int base = (f4 << 2) * stride;
          const int s0 = lds_pad_slot(base + (int)b4.x);
          const int s1 = lds_pad_slot(base + stride + (int)b4.y);
          const int s2 = lds_pad_slot(base + 2 * stride + (int)b4.z);
          const int s3 = lds_pad_slot(base + 3 * stride + (int)b4.w);
          atomicAdd(&lhist[s0], gfix);
          atomicAdd(&lhist[hofs + s0], hfix);
          atomicAdd(&lhist[s1], gfix);
          atomicAdd(&lhist[hofs + s1], hfix);
          atomicAdd(&lhist[s2], gfix);
          atomicAdd(&lhist[hofs + s2], hfix);
          atomicAdd(&lhist[s3], gfix);
          atomicAdd(&lhist[hofs + s3], hfix);
        }
        continue;
      }
      #pragma unroll 4
      for (int f = 0; f < nf_group; ++f) {
        const int slot2 = lds_pad_slot(f * stride + (int)rp[f]);
        atomicAdd(&lhist[slot2], gfix);
        atomicAdd(&lhist[hofs + slot2], hfix);
      }
    }
    __syncthreads();
    unsigned long long* gout =
        out + ((long long)slot * nfeat + fg_start) * (long long)stride * 2;
    for (int i = threadIdx.x; i < lds_words; i += blockDim.x) {
      const int pair = i >> 1;
      const int idx = lds_pad_slot(pair) + ((i & 1) ? hofs : 0);
      const unsigned long long v = lhist[idx];
      if (v) atomicAdd(&gout[i], v);
    }
    __syncthreads();
  }
}

// convert built slots (int64 acc -> f32 heap) and fill node sums from the
// feature-0 bin range; one block per (slot, chunk of slots_total)
__global__ __launch_bounds__(HIST_BLOCK) void convert_level_kernel(
    const unsigned long long* __restrict__ acc, float* __restrict__ hist_f32,
    const LevelNode* __restrict__ nodes, int k, long long slots2,
    const float* __restrict__ gh_max) {
  const double inv_g = (double)fmaxf(gh_max[0], 1e-30f) / 8589934592.0;
  const double inv_h = (double)fmaxf(gh_max[1], 1e-30f) / 8589934592.0;
  for (long long u = (long long)blockIdx.x * blockDim.x + threadIdx.x; u < (long long)k * slots2;
       u += (long long)gridDim.x * blockDim.x) {
    const int slot = (int)(u / slots2);
    if (!nodes[slot].build || nodes[slot].end <= nodes[slot].start) continue;
    const long long j = u - (long long)slot * slots2;
    const double inv = (j & 1) ? inv_h : inv_g;
    hist_f32[u] = (float)((double)(long long)acc[u] * inv);
  }
}

// derived slots: hist = parent - sibling (all f32, same heap layout);
// also node sums for EVERY alive slot from the feature-0 bins
__global__ __launch_bounds__(HIST_BLOCK) void derive_level_kernel(
    float* __restrict__ level_hist, const float* __restrict__ parent_hist,
    const LevelNode* __restrict__ nodes, int k, long long slots2) {
  for (long long u = (long long)blockIdx.x * blockDim.x + threadIdx.x; u < (long long)k * slots2;
       u += (long long)gridDim.x * blockDim.x) {
    const int slot = (int)(u / slots2);
    const LevelNode node = nodes[slot];
    if (node.build || node.end <= node.start) continue;
    const long long j = u - (long long)slot * slots2;
    const int sib = slot ^ 1;
    const int parent = slot >> 1;
    level_hist[u] = parent_hist[(long long)parent * slots2 + j] -
                    level_hist[(long long)sib * slots2 + j];
  }
}

// LDS-staged partition variant: each tile's payload (bins rows, gh, row
// ids) is read from HBM exactly ONCE into LDS during classification, and
// the scatter writes read LDS instead of re-pulling the source through L2
// — removes the second source pass (~28 B/row/level on Higgs shape).
// u8 bins with nfeat % 4 == 0 only; dynamic LDS = tile*(8 + nd*4 + 8) B.
template <int TILE>
__global__ __launch_bounds__(HIST_BLOCK) void partition_device_lds_kernel(
    const unsigned char* __restrict__ src_bins, const float2* __restrict__ src_gh,
    const int* __restrict__ src_rows, unsigned char* __restrict__ dst_bins,
    float2* __restrict__ dst_gh, int* __restrict__ dst_rows,
    const LevelNode* __restrict__ nodes, const int* __restrict__ part_prefix,
    const LevelWork* __restrict__ work, const float* __restrict__ split_packed,
    int* __restrict__ counters, int k, int nfeat, int missing_bin,
    int copy_payload) {
  extern __shared__ unsigned char smem[];
  const int nd = nfeat >> 2;
  float2* lgh = reinterpret_cast<float2*>(smem);                       // [TILE]
  uchar4* lbins = reinterpret_cast<uchar4*>(smem + TILE * 8);          // [TILE*nd]
  int* lrows = reinterpret_cast<int*>(smem + TILE * 8 + (size_t)TILE * nd * 4);
  int* ldest = lrows + TILE;
  __shared__ int lcnt, rcnt, lbase, rbase;
  const int total = work->part_total;

  for (int vb = blockIdx.x; vb < total; vb += gridDim.x) {
    const int slot = find_slot(part_prefix, k, vb);
    const LevelNode node = nodes[slot];
    const float* sp6 = split_packed + slot * 6;
    if (sp6[0] <= 0.0f || node.end <= node.start) continue;
    const int feature = (int)sp6[1];
    const int split_bin = (int)sp6[2];
    const int default_left = sp6[3] > 0.5f ? 1 : 0;
    const int chunk = vb - part_prefix[slot];
    const int nb = part_prefix[slot + 1] - part_prefix[slot];
    const long long tile_step = (long long)nb * TILE;

    for (long long tile = node.start + (long long)chunk * TILE; tile < node.end;
         tile += tile_step) {
      if (threadIdx.x == 0) {
        lcnt = 0;
        rcnt = 0;
      }
      __syncthreads();
      const int tile_n = (int)min((long long)TILE, node.end - tile);
      // single coalesced source pass: bins -> LDS (vec4), gh + rows -> LDS
      const uchar4* sb4 = reinterpret_cast<const uchar4*>(src_bins);
      for (int u = threadIdx.x; u < tile_n * nd; u += blockDim.x) {
        lbins[u] = sb4[(tile + (u / nd)) * (long long)nd + (u % nd)];
      }
      for (int i = threadIdx.x; i < tile_n; i += blockDim.x) {
        lgh[i] = src_gh[tile + i];
        lrows[i] = src_rows[tile + i];
      }
      __syncthreads();
      for (int i = threadIdx.x; i < tile_n; i += blockDim.x) {
        const int b =
            (int)reinterpret_cast<const unsigned char*>(lbins)[i * (long long)nfeat + feature];
        const bool left = (b == missing_bin) ? (default_left != 0) : (b <= split_bin);
        ldest[i] = left ? atomicAdd(&lcnt, 1) : ~atomicAdd(&rcnt, 1);
      }
      __syncthreads();
      if (threadIdx.x == 0) {
        lbase = atomicAdd(&counters[slot * 2], lcnt);
        rbase = atomicAdd(&counters[slot * 2 + 1], rcnt);
      }
      __syncthreads();
      if (copy_payload) {
        int log2p = 0;
        while ((1 << log2p) < nd) ++log2p;
        const int mask = (1 << log2p) - 1;
        uchar4* db4 = reinterpret_cast<uchar4*>(dst_bins);
        for (int u = threadIdx.x; u < (tile_n << log2p); u += blockDim.x) {
          const int i = u >> log2p;
          const int f4 = u & mask;
          if (f4 >= nd) continue;
          const int d = ldest[i];
          const long long dst =
              d >= 0 ? (long long)node.start + lbase + d : (long long)node.end - 1 - rbase - (~d);
          db4[dst * nd + f4] = lbins[i * nd + f4];
        }
        for (int i = threadIdx.x; i < tile_n; i += blockDim.x) {
          const int d = ldest[i];
          const long long dst =
              d >= 0 ? (long long)node.start + lbase + d : (long long)node.end - 1 - rbase - (~d);
          dst_gh[dst] = lgh[i];
        }
      }
      for (int i = threadIdx.x; i < tile_n; i += blockDim.x) {
        const int d = ldest[i];
        const long long dst =
            d >= 0 ? (long long)node.start + lbase + d : (long long)node.end - 1 - rbase - (~d);
        dst_rows[dst] = lrows[i];
      }
      __syncthreads();
    }
  }
}

// partition driven by the device job table (virtual blocks)
template <typename BinT>
__global__ __launch_bounds__(HIST_BLOCK) void partition_device_kernel(
    const BinT* __restrict__ src_bins, const float2* __restrict__ src_gh,
    const int* __restrict__ src_rows, BinT* __restrict__ dst_bins,
    float2* __restrict__ dst_gh, int* __restrict__ dst_rows,
    const LevelNode* __restrict__ nodes, const int* __restrict__ part_prefix,
    const LevelWork* __restrict__ work, const float* __restrict__ split_packed,
    int* __restrict__ counters, int k, int nfeat, int missing_bin,
    int copy_payload) {
  __shared__ int ldest[CPART_TILE];
  __shared__ int lcnt, rcnt, lbase, rbase;
  const int total = work->part_total;

  for (int vb = blockIdx.x; vb < total; vb += gridDim.x) {
    const int slot = find_slot(part_prefix, k, vb);
    const LevelNode node = nodes[slot];
    const float* sp6 = split_packed + slot * 6;
    if (sp6[0] <= 0.0f || node.end <= node.start) continue;
    const int feature = (int)sp6[1];
    const int split_bin = (int)sp6[2];
    const int default_left = sp6[3] > 0.5f ? 1 : 0;
    const int chunk = vb - part_prefix[slot];
    const int nb = part_prefix[slot + 1] - part_prefix[slot];
    const long long tile_step = (long long)nb * CPART_TILE;

    for (long long tile = node.start + (long long)chunk * CPART_TILE; tile < node.end;
         tile += tile_step) {
      if (threadIdx.x == 0) {
        lcnt = 0;
        rcnt = 0;
      }
      __syncthreads();
      const int tile_n = (int)min((long long)CPART_TILE, node.end - tile);
      for (int i = threadIdx.x; i < tile_n; i += blockDim.x) {
        const int b = (int)src_bins[(tile + i) * (long long)nfeat + feature];
        const bool left = (b == missing_bin) ? (default_left != 0) : (b <= split_bin);
        ldest[i] = left ? atomicAdd(&lcnt, 1) : ~atomicAdd(&rcnt, 1);
      }
      __syncthreads();
      if (threadIdx.x == 0) {
        lbase = atomicAdd(&counters[slot * 2], lcnt);
        rbase = atomicAdd(&counters[slot * 2 + 1], rcnt);
      }
      __syncthreads();
      // the deepest level's partition only feeds leaf_update (row ids +
      // counts): bins/gh copies are skipped there (copy_payload == 0),
      // dropping ~90% of that level's partition traffic.
      if (copy_payload && (nfeat & 3) == 0 && sizeof(BinT) == 1) {
        const int nd = nfeat >> 2;
        int log2p = 0;
        while ((1 << log2p) < nd) ++log2p;
        const int mask = (1 << log2p) - 1;
        const uchar4* sb4 = reinterpret_cast<const uchar4*>(src_bins);
        uchar4* db4 = reinterpret_cast<uchar4*>(dst_bins);
        for (int u = threadIdx.x; u < (tile_n << log2p); u += blockDim.x) {
          const int i = u >> log2p;
          const int f4 = u & mask;
          if (f4 >= nd) continue;
          const int d = ldest[i];
          const long long dst =
              d >= 0 ? (long long)node.start + lbase + d : (long long)node.end - 1 - rbase - (~d);
          db4[dst * nd + f4] = sb4[(tile + i) * (long long)nd + f4];
        }
      } else if (copy_payload) {
        int log2p = 0;
        while ((1 << log2p) < nfeat) ++log2p;
        const int mask = (1 << log2p) - 1;
        for (int u = threadIdx.x; u < (tile_n << log2p); u += blockDim.x) {
          const int i = u >> log2p;
          const int f = u & mask;
          if (f >= nfeat) continue;
          const int d = ldest[i];
          const long long dst =
              d >= 0 ? (long long)node.start + lbase + d : (long long)node.end - 1 - rbase - (~d);
          dst_bins[dst * (long long)nfeat + f] = src_bins[(tile + i) * (long long)nfeat + f];
        }
      }
      if (copy_payload) {
        for (int u = threadIdx.x; u < tile_n * 2; u += blockDim.x) {
          const int i = u >> 1;
          const int half = u & 1;
          const int d = ldest[i];
          const long long dst =
              d >= 0 ? (long long)node.start + lbase + d : (long long)node.end - 1 - rbase - (~d);
          reinterpret_cast<float*>(dst_gh)[dst * 2 + half] =
              reinterpret_cast<const float*>(src_gh)[(tile + i) * 2 + half];
        }
      }
      for (int i = threadIdx.x; i < tile_n; i += blockDim.x) {
        const int d = ldest[i];
        const long long dst =
            d >= 0 ? (long long)node.start + lbase + d : (long long)node.end - 1 - rbase - (~d);
        dst_rows[dst] = src_rows[tile + i];
      }
      __syncthreads();
    }
  }
}

// ---------------------------------------------------------------------------
// split-gain scan
//
// Kernel A: one 256-thread workgroup per (node, feature): inclusive scan of
// the feature's bin histogram in LDS, gain for every split position in both
// missing directions, block-reduce to the feature's best candidate.
// Kernel B: one workgroup per node reduces over features.
// Replaces ~30 small torch kernels + 6 D2H syncs per level.
// ---------------------------------------------------------------------------

struct SplitCand {
  float gain;
  int bin;
  int dir;  // 1 = missing left
  float left_g;
  float left_h;
};

__device__ inline float split_score(float g, float h, float alpha, float lam) {
  float ag = fabsf(g) - alpha;
  ag = ag > 0.f ? ag : 0.f;
  return ag * ag / (h + lam);
}

__device__ inline float split_weight(float g, float h, float alpha, float lam) {
  float ag = fabsf(g) - alpha;
  ag = ag > 0.f ? ag : 0.f;
  return -copysignf(ag, g) / (h + lam);
}

#define SPLIT_BLOCK 256

__global__ __launch_bounds__(SPLIT_BLOCK) void split_scan_kernel(
    const float* __restrict__ hist,      // [k, f, stride, 2]
    const float2* __restrict__ parent,   // [k]
    const int* __restrict__ nbins,       // [f]
    const unsigned char* __restrict__ feat_mask,  // [k*f], [f] or null
    const signed char* __restrict__ monotone,     // [f] or null
    SplitCand* __restrict__ out,         // [k, f]
    int k, int f, int stride, int has_missing, int mask_per_node,
    float reg_lambda, float reg_alpha, float gamma_, float min_child_weight) {
  __shared__ float sg[SPLIT_BLOCK];
  __shared__ float sh[SPLIT_BLOCK];
  __shared__ float red_gain[SPLIT_BLOCK / WAVE];
  __shared__ int red_idx[SPLIT_BLOCK / WAVE];

  const int node = blockIdx.x / f;
  const int feat = blockIdx.x % f;
  const int tid = threadIdx.x;
  SplitCand best = {-1.0f, -1, 0, 0.f, 0.f};

  bool masked = false;
  if (feat_mask != nullptr) {
    masked = feat_mask[mask_per_node ? (node * f + feat) : feat] == 0;
  }
  const int nb = nbins[feat];  // real bins
  if (!masked && nb >= 2 && nb <= SPLIT_BLOCK) {
    const float* hbase = hist + (((long long)node * f + feat) * stride) * 2;
    const float2 psum = parent[node];

    // load + inclusive block scan of real bins (nb <= 256)
    float g = 0.f, h = 0.f;
    if (tid < nb) {
      g = hbase[tid * 2];
      h = hbase[tid * 2 + 1];
    }
    // scan in LDS (Hillis-Steele; nb small)
    sg[tid] = g;
    sh[tid] = h;
    __syncthreads();
    for (int ofs = 1; ofs < nb; ofs <<= 1) {
      float ag = 0.f, ah = 0.f;
      if (tid >= ofs) {
        ag = sg[tid - ofs];
        ah = sh[tid - ofs];
      }
      __syncthreads();
      sg[tid] += ag;
      sh[tid] += ah;
      __syncthreads();
    }

    float miss_g = 0.f, miss_h = 0.f;
    if (has_missing) {
      miss_g = hbase[(stride - 1) * 2];
      miss_h = hbase[(stride - 1) * 2 + 1];
    }
    const float parent_score = split_score(psum.x, psum.y, reg_alpha, reg_lambda);
    const int cons = (monotone != nullptr) ? (int)monotone[feat] : 0;

    // split after bin j valid for j in [0, nb-2]
    if (tid <= nb - 2) {
      const float gl0 = sg[tid];
      const float hl0 = sh[tid];
      for (int dir = 0; dir < 2; ++dir) {
        const float gl = gl0 + (dir ? miss_g : 0.f);
        const float hl = hl0 + (dir ? miss_h : 0.f);
        const float gr = psum.x - gl;
        const float hr = psum.y - hl;
        if (hl < min_child_weight || hr < min_child_weight) continue;
        if (cons != 0) {
          const float wl = split_weight(gl, hl, reg_alpha, reg_lambda);
          const float wr = split_weight(gr, hr, reg_alpha, reg_lambda);
          if ((cons > 0 && wl > wr) || (cons < 0 && wl < wr)) continue;
        }
        const float gain =
            0.5f * (split_score(gl, hl, reg_alpha, reg_lambda) +
                    split_score(gr, hr, reg_alpha, reg_lambda) - parent_score) - gamma_;
        if (gain > best.gain) {
          best = {gain, tid, dir, gl, hl};
        }
      }
    }
  }
  __syncthreads();

  // block argmax reduce over candidates (pack gain+lane via wave shuffle)
  float bg = best.gain;
  int bidx = tid;
  for (int ofs = WAVE / 2; ofs > 0; ofs >>= 1) {
    const float og = __shfl_down(bg, ofs);
    const int oi = __shfl_down(bidx, ofs);
    if (og > bg) {
      bg = og;
      bidx = oi;
    }
  }
  const int wid = tid / WAVE;
  if ((tid & (WAVE - 1)) == 0) {
    red_gain[wid] = bg;
    red_idx[wid] = bidx;
  }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < SPLIT_BLOCK / WAVE; ++w) {
      if (red_gain[w] > red_gain[0]) {
        red_gain[0] = red_gain[w];
        red_idx[0] = red_idx[w];
      }
    }
  }
  __syncthreads();
  // winning thread writes its candidate
  if (tid == red_idx[0] && best.gain == red_gain[0]) {
    out[(long long)node * f + feat] = best;
  }
}

__global__ __launch_bounds__(SPLIT_BLOCK) void split_reduce_kernel(
    const SplitCand* __restrict__ cands,  // [k, f]
    float* __restrict__ out,              // [k, 6]: gain, feat, bin, dir, lg, lh
    int k, int f) {
  __shared__ float rg[SPLIT_BLOCK / WAVE];
  __shared__ int ri[SPLIT_BLOCK / WAVE];
  const int node = blockIdx.x;
  const int tid = threadIdx.x;
  float bg = -1.0f;
  int bf = -1;
  for (int j = tid; j < f; j += blockDim.x) {
    const float gn = cands[(long long)node * f + j].gain;
    if (gn > bg) {
      bg = gn;
      bf = j;
    }
  }
  for (int ofs = WAVE / 2; ofs > 0; ofs >>= 1) {
    const float og = __shfl_down(bg, ofs);
    const int of_ = __shfl_down(bf, ofs);
    if (og > bg) {
      bg = og;
      bf = of_;
    }
  }
  const int wid = tid / WAVE;
  if ((tid & (WAVE - 1)) == 0) {
    rg[wid] = bg;
    ri[wid] = bf;
  }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < SPLIT_BLOCK / WAVE; ++w) {
      if (rg[w] > rg[0]) {
        rg[0] = rg[w];
        ri[0] = ri[w];
      }
    }
    float* o = out + node * 6;
    if (ri[0] < 0 || rg[0] <= 0.f) {
      o[0] = -1.0f;
      o[1] = -1.f;
      o[2] = -1.f;
      o[3] = 0.f;
      o[4] = 0.f;
      o[5] = 0.f;
    } else {
      const SplitCand c = cands[(long long)node * f + ri[0]];
      o[0] = c.gain;
      o[1] = (float)ri[0];
      o[2] = (float)c.bin;
      o[3] = (float)c.dir;
      o[4] = c.left_g;
      o[5] = c.left_h;
    }
  }
}

// ---------------------------------------------------------------------------
// row partition (two-ended compaction within each segment)
// ---------------------------------------------------------------------------

// Rows are staged through LDS tiles so each block issues ONE global atomic
// per side per tile (vs one per row: a single counter word sustains only
// ~88 atomics/us even wave-aggregated — measured 2.9 ms/level before).
#define PART_TILE 4096

template <typename BinT>
__global__ __launch_bounds__(HIST_BLOCK) void partition_kernel(
    const BinT* __restrict__ bins, const int* __restrict__ src, int* __restrict__ dst,
    const PartJob* __restrict__ jobs, const int* __restrict__ block_job,
    int* __restrict__ counters, int nfeat, int missing_bin) {
  __shared__ int lbuf[PART_TILE];
  __shared__ int rbuf[PART_TILE];
  __shared__ int lcnt, rcnt, lbase, rbase;

  const int j = block_job[blockIdx.x];
  const PartJob job = jobs[j];
  const int chunk = blockIdx.x - job.first_block;
  const long long tile_step = (long long)job.num_blocks * PART_TILE;

  for (long long tile = job.start + (long long)chunk * PART_TILE; tile < job.end; tile += tile_step) {
    if (threadIdx.x == 0) {
      lcnt = 0;
      rcnt = 0;
    }
    __syncthreads();
    const int tile_n = (int)min((long long)PART_TILE, job.end - tile);
    for (int i = threadIdx.x; i < tile_n; i += blockDim.x) {
      const int row = src[tile + i];
      const int b = (int)bins[(long long)row * nfeat + job.feature];
      const bool left = (b == missing_bin) ? (job.default_left != 0) : (b <= job.split_bin);
      if (left) {
        lbuf[atomicAdd(&lcnt, 1)] = row;
      } else {
        rbuf[atomicAdd(&rcnt, 1)] = row;
      }
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      lbase = atomicAdd(&counters[j * 2], lcnt);
      rbase = atomicAdd(&counters[j * 2 + 1], rcnt);
    }
    __syncthreads();
    for (int i = threadIdx.x; i < lcnt; i += blockDim.x) dst[job.start + lbase + i] = lbuf[i];
    for (int i = threadIdx.x; i < rcnt; i += blockDim.x) dst[job.end - 1 - rbase - i] = rbuf[i];
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// leaf value scatter into the margin vector
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(HIST_BLOCK) void leaf_update_kernel(
    const int* __restrict__ buf0, const int* __restrict__ buf1,
    float* __restrict__ margin, const LeafJob* __restrict__ jobs,
    const int* __restrict__ block_job, long long col_stride) {
  const LeafJob job = jobs[block_job[blockIdx.x]];
  const int* src = job.parity ? buf1 : buf0;
  const int chunk = blockIdx.x - job.first_block;
  const long long step = (long long)job.num_blocks * blockDim.x;
  for (long long r = job.start + (long long)chunk * blockDim.x + threadIdx.x; r < job.end; r += step) {
    margin[(long long)src[r] * col_stride] += job.value;
  }
}

// ---------------------------------------------------------------------------
// batched forest prediction (dense rows x trees traversal)
// ---------------------------------------------------------------------------

// Work item = (row, tree-chunk): small serving batches (1k rows) would
// otherwise launch ~4 blocks against 256 CUs and serialize 500 trees per
// thread (measured 1.36 ms for 1k x 500). Each (row, chunk) owns its
// exclusive out[chunk, row, :] slice — no atomics; the caller reduces the
// chunk axis with a fixed-order torch sum (deterministic).
__global__ __launch_bounds__(HIST_BLOCK) void predict_kernel(
    const float* __restrict__ X, long long n, int nfeat,
    const int* __restrict__ left, const int* __restrict__ right,
    const int* __restrict__ feat, const float* __restrict__ thresh,
    const unsigned char* __restrict__ defl, const float* __restrict__ value,
    const int* __restrict__ tree_root, const int* __restrict__ tree_cls,
    int t_begin, int t_end, float* __restrict__ out, int k, int n_chunks,
    int trees_per_chunk) {
  const long long total = n * (long long)n_chunks;
  const long long step = (long long)gridDim.x * blockDim.x;
  for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += step) {
    const long long row = idx / n_chunks;
    const int c = (int)(idx - row * n_chunks);
    const float* xr = X + row * nfeat;
    float* orow = out + ((long long)c * n + row) * k;
    const int ts = t_begin + c * trees_per_chunk;
    const int te = min(t_end, ts + trees_per_chunk);
    for (int t = ts; t < te; ++t) {
      int nid = tree_root[t];
      int l;
      while ((l = left[nid]) >= 0) {
        const float fv = xr[feat[nid]];
        const bool goleft = isnan(fv) ? (defl[nid] != 0) : (fv < thresh[nid]);
        nid = goleft ? l : right[nid];
      }
      orow[tree_cls[t]] += value[nid];
    }
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

static hipStream_t current_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

void hist_build(torch::Tensor bins, torch::Tensor gh, torch::Tensor rowbuf,
                torch::Tensor jobs, torch::Tensor block_job, torch::Tensor out,
                int64_t nfeat, int64_t stride, double scale_g, double scale_h,
                int64_t lds_words) {
  CHECK_GPU(bins);
  CHECK_GPU(out);
  const int grid = (int)block_job.size(0);
  const size_t lds_bytes = (size_t)lds_words * sizeof(unsigned long long);
  auto stream = current_stream();
  if (bins.scalar_type() == torch::kUInt8) {
    hipLaunchKernelGGL(hist_kernel<unsigned char>, dim3(grid), dim3(HIST_BLOCK), lds_bytes, stream,
                       bins.data_ptr<unsigned char>(), (const float2*)gh.data_ptr<float>(),
                       rowbuf.data_ptr<int>(), (const HistJob*)jobs.data_ptr<int>(),
                       block_job.data_ptr<int>(), (unsigned long long*)out.data_ptr<int64_t>(),
                       (int)nfeat, (int)stride, (float)scale_g, (float)scale_h);
  } else {
    hipLaunchKernelGGL(hist_kernel<short>, dim3(grid), dim3(HIST_BLOCK), lds_bytes, stream,
                       bins.data_ptr<short>(), (const float2*)gh.data_ptr<float>(),
                       rowbuf.data_ptr<int>(), (const HistJob*)jobs.data_ptr<int>(),
                       block_job.data_ptr<int>(), (unsigned long long*)out.data_ptr<int64_t>(),
                       (int)nfeat, (int)stride, (float)scale_g, (float)scale_h);
  }
}

void hist_convert(torch::Tensor in, torch::Tensor out, double inv_g, double inv_h) {
  CHECK_GPU(in);
  const long long n_pairs = in.numel() / 2;
  const int grid = (int)std::min<long long>((n_pairs + HIST_BLOCK - 1) / HIST_BLOCK, 2048);
  hipLaunchKernelGGL(hist_convert_kernel, dim3(std::max(grid, 1)), dim3(HIST_BLOCK), 0, current_stream(),
                     (const unsigned long long*)in.data_ptr<int64_t>(), out.data_ptr<float>(),
                     n_pairs, (float)inv_g, (float)inv_h);
}

void partition(torch::Tensor bins, torch::Tensor src, torch::Tensor dst,
               torch::Tensor jobs, torch::Tensor block_job, torch::Tensor counters,
               int64_t nfeat, int64_t missing_bin) {
  CHECK_GPU(bins);
  const int grid = (int)block_job.size(0);
  auto stream = current_stream();
  if (bins.scalar_type() == torch::kUInt8) {
    hipLaunchKernelGGL(partition_kernel<unsigned char>, dim3(grid), dim3(HIST_BLOCK), 0, stream,
                       bins.data_ptr<unsigned char>(), src.data_ptr<int>(), dst.data_ptr<int>(),
                       (const PartJob*)jobs.data_ptr<int>(), block_job.data_ptr<int>(),
                       counters.data_ptr<int>(), (int)nfeat, (int)missing_bin);
  } else {
    hipLaunchKernelGGL(partition_kernel<short>, dim3(grid), dim3(HIST_BLOCK), 0, stream,
                       bins.data_ptr<short>(), src.data_ptr<int>(), dst.data_ptr<int>(),
                       (const PartJob*)jobs.data_ptr<int>(), block_job.data_ptr<int>(),
                       counters.data_ptr<int>(), (int)nfeat, (int)missing_bin);
  }
}

void leaf_update(torch::Tensor buf0, torch::Tensor buf1, torch::Tensor margin_base,
                 torch::Tensor jobs, torch::Tensor block_job, int64_t col_stride) {
  CHECK_GPU(margin_base);
  const int grid = (int)block_job.size(0);
  hipLaunchKernelGGL(leaf_update_kernel, dim3(grid), dim3(HIST_BLOCK), 0, current_stream(),
                     buf0.data_ptr<int>(), buf1.data_ptr<int>(), margin_base.data_ptr<float>(),
                     (const LeafJob*)jobs.data_ptr<int>(), block_job.data_ptr<int>(), col_stride);
}

void predict_forest(torch::Tensor X, torch::Tensor left, torch::Tensor right,
                    torch::Tensor feat, torch::Tensor thresh, torch::Tensor defl,
                    torch::Tensor value, torch::Tensor tree_root, torch::Tensor tree_cls,
                    int64_t t_begin, int64_t t_end, torch::Tensor out, int64_t k,
                    int64_t n_chunks, int64_t trees_per_chunk) {
  CHECK_GPU(X);
  const long long n = X.size(0);
  const long long total = n * n_chunks;
  const int grid = (int)std::min<long long>((total + HIST_BLOCK - 1) / HIST_BLOCK, 4096);
  hipLaunchKernelGGL(predict_kernel, dim3(std::max(grid, 1)), dim3(HIST_BLOCK), 0, current_stream(),
                     X.data_ptr<float>(), n, (int)X.size(1), left.data_ptr<int>(),
                     right.data_ptr<int>(), feat.data_ptr<int>(), thresh.data_ptr<float>(),
                     defl.data_ptr<unsigned char>(), value.data_ptr<float>(),
                     tree_root.data_ptr<int>(), tree_cls.data_ptr<int>(),
                     (int)t_begin, (int)t_end, out.data_ptr<float>(), (int)k, (int)n_chunks,
                     (int)trees_per_chunk);
}

void hist_build_compact(torch::Tensor bins_c, torch::Tensor gh_c, torch::Tensor jobs,
                        torch::Tensor block_job, torch::Tensor out, int64_t nfeat,
                        int64_t stride, torch::Tensor gh_max, int64_t lds_words) {
  CHECK_GPU(bins_c);
  const int grid = (int)block_job.size(0);
  const size_t lds_bytes = (size_t)lds_words * sizeof(unsigned long long);
  auto stream = current_stream();
  if (bins_c.scalar_type() == torch::kUInt8) {
    hipLaunchKernelGGL(hist_compact_kernel<unsigned char>, dim3(grid), dim3(HIST_BLOCK), lds_bytes,
                       stream, bins_c.data_ptr<unsigned char>(),
                       (const float2*)gh_c.data_ptr<float>(), (const HistJob*)jobs.data_ptr<int>(),
                       block_job.data_ptr<int>(), (unsigned long long*)out.data_ptr<int64_t>(),
                       (int)nfeat, (int)stride, gh_max.data_ptr<float>());
  } else {
    hipLaunchKernelGGL(hist_compact_kernel<short>, dim3(grid), dim3(HIST_BLOCK), lds_bytes, stream,
                       bins_c.data_ptr<short>(), (const float2*)gh_c.data_ptr<float>(),
                       (const HistJob*)jobs.data_ptr<int>(), block_job.data_ptr<int>(),
                       (unsigned long long*)out.data_ptr<int64_t>(), (int)nfeat, (int)stride,
                       gh_max.data_ptr<float>());
  }
}

void hist_convert_dev(torch::Tensor in, torch::Tensor out, torch::Tensor gh_max) {
  CHECK_GPU(in);
  const long long n_pairs = in.numel() / 2;
  const int grid = (int)std::min<long long>((n_pairs + HIST_BLOCK - 1) / HIST_BLOCK, 2048);
  hipLaunchKernelGGL(hist_convert_dev_kernel, dim3(std::max(grid, 1)), dim3(HIST_BLOCK), 0,
                     current_stream(), (const unsigned long long*)in.data_ptr<int64_t>(),
                     out.data_ptr<float>(), n_pairs, gh_max.data_ptr<float>());
}

void partition_compact(torch::Tensor src_bins, torch::Tensor src_gh, torch::Tensor src_rows,
                       torch::Tensor dst_bins, torch::Tensor dst_gh, torch::Tensor dst_rows,
                       torch::Tensor jobs, torch::Tensor block_job, torch::Tensor split_packed,
                       torch::Tensor counters, int64_t nfeat, int64_t missing_bin) {
  CHECK_GPU(src_bins);
  const int grid = (int)block_job.size(0);
  auto stream = current_stream();
  if (src_bins.scalar_type() == torch::kUInt8) {
    hipLaunchKernelGGL(partition_compact_kernel<unsigned char>, dim3(grid), dim3(HIST_BLOCK), 0,
                       stream, src_bins.data_ptr<unsigned char>(),
                       (const float2*)src_gh.data_ptr<float>(), src_rows.data_ptr<int>(),
                       dst_bins.data_ptr<unsigned char>(), (float2*)dst_gh.data_ptr<float>(),
                       dst_rows.data_ptr<int>(), (const PartJob*)jobs.data_ptr<int>(),
                       block_job.data_ptr<int>(), split_packed.data_ptr<float>(),
                       counters.data_ptr<int>(), (int)nfeat, (int)missing_bin);
  } else {
    hipLaunchKernelGGL(partition_compact_kernel<short>, dim3(grid), dim3(HIST_BLOCK), 0, stream,
                       src_bins.data_ptr<short>(), (const float2*)src_gh.data_ptr<float>(),
                       src_rows.data_ptr<int>(), dst_bins.data_ptr<short>(),
                       (float2*)dst_gh.data_ptr<float>(), dst_rows.data_ptr<int>(),
                       (const PartJob*)jobs.data_ptr<int>(), block_job.data_ptr<int>(),
                       split_packed.data_ptr<float>(), counters.data_ptr<int>(), (int)nfeat,
                       (int)missing_bin);
  }
}

void leaf_update_compact(torch::Tensor rows0, torch::Tensor rows1, torch::Tensor margin_base,
                         torch::Tensor jobs, torch::Tensor block_job, int64_t col_stride) {
  CHECK_GPU(margin_base);
  const int grid = (int)block_job.size(0);
  hipLaunchKernelGGL(leaf_update_compact_kernel, dim3(grid), dim3(HIST_BLOCK), 0, current_stream(),
                     rows0.data_ptr<int>(), rows1.data_ptr<int>(), margin_base.data_ptr<float>(),
                     (const LeafJob*)jobs.data_ptr<int>(), block_job.data_ptr<int>(), col_stride);
}

void grow_make_root(torch::Tensor nodes, torch::Tensor hist_prefix, torch::Tensor part_prefix,
                    torch::Tensor work, int64_t cap, int64_t rows_per_block, int64_t max_blocks) {
  hipLaunchKernelGGL(make_root_kernel, dim3(1), dim3(1), 0, current_stream(),
                     (LevelNode*)nodes.data_ptr<int>(), hist_prefix.data_ptr<int>(),
                     part_prefix.data_ptr<int>(), (LevelWork*)work.data_ptr<int>(), (int)cap,
                     (int)rows_per_block, (int)max_blocks);
}

void grow_make_level(torch::Tensor cur, torch::Tensor splits, torch::Tensor counts,
                     torch::Tensor node_gh, torch::Tensor nxt, torch::Tensor nxt_gh,
                     torch::Tensor hist_prefix, torch::Tensor part_prefix, torch::Tensor work,
                     int64_t k, int64_t rows_per_block, int64_t max_blocks) {
  hipLaunchKernelGGL(make_level_kernel, dim3(1), dim3(GROW_MAX_SLOTS), 0, current_stream(),
                     (const LevelNode*)cur.data_ptr<int>(), splits.data_ptr<float>(),
                     counts.data_ptr<int>(), node_gh.data_ptr<float>(),
                     (LevelNode*)nxt.data_ptr<int>(), nxt_gh.data_ptr<float>(),
                     hist_prefix.data_ptr<int>(), part_prefix.data_ptr<int>(),
                     (LevelWork*)work.data_ptr<int>(), (int)k, (int)rows_per_block,
                     (int)max_blocks);
}

// dispatch helper: BLOCK is a compile-time launch bound (256 grouped-slab,
// 512 single full-feature slab)
template <typename BinT>
static void launch_hist_device(int grid, int hist_block, size_t lds_bytes, hipStream_t stream,
                               const BinT* bins_c, const float2* gh_c, const LevelNode* nodes,
                               const int* hist_prefix, const LevelWork* work,
                               unsigned long long* acc, int k, int nfeat, int stride,
                               int n_groups, int feats_per_group, const float* gh_max,
                               int rows_per_block, int slot_lo = 0, int slot_hi = 1 << 30) {
  // hipGraph kernel nodes validate dynamic LDS against the function's
  // max-dynamic-shared attribute (plain launches do not): without this
  // opt-in a captured full-slab launch (>64 KB) reads a truncated LDS
  // allocation and memory-faults on replay (gfx950 allows 160 KB).
  static bool lds_attr_set = [] {
    (void)hipFuncSetAttribute((const void*)(hist_device_kernel<BinT, 512>),
                              hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
    (void)hipFuncSetAttribute((const void*)(hist_device_kernel<BinT, HIST_BLOCK>),
                              hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
    return true;
  }();
  (void)lds_attr_set;
  if (hist_block == 512) {
    hipLaunchKernelGGL((hist_device_kernel<BinT, 512>), dim3(grid), dim3(512), lds_bytes, stream,
                       bins_c, gh_c, nodes, hist_prefix, work, acc, k, nfeat, stride, n_groups,
                       feats_per_group, gh_max, rows_per_block, slot_lo, slot_hi);
  } else {
    hipLaunchKernelGGL((hist_device_kernel<BinT, HIST_BLOCK>), dim3(grid), dim3(HIST_BLOCK),
                       lds_bytes, stream, bins_c, gh_c, nodes, hist_prefix, work, acc, k, nfeat,
                       stride, n_groups, feats_per_group, gh_max, rows_per_block, slot_lo,
                       slot_hi);
  }
}

void grow_hist_level(torch::Tensor bins_c, torch::Tensor gh_c, torch::Tensor nodes,
                     torch::Tensor hist_prefix, torch::Tensor work, torch::Tensor acc,
                     int64_t k, int64_t nfeat, int64_t stride, int64_t n_groups,
                     int64_t feats_per_group, torch::Tensor gh_max, int64_t rows_per_block,
                     int64_t grid, int64_t lds_words, int64_t hist_block,
                     int64_t slot_lo, int64_t slot_hi) {
  const size_t lds_bytes = (size_t)lds_words * sizeof(unsigned long long);
  auto stream = current_stream();
  if (bins_c.scalar_type() == torch::kUInt8) {
    launch_hist_device<unsigned char>(
        (int)grid, (int)hist_block, lds_bytes, stream, bins_c.data_ptr<unsigned char>(),
        (const float2*)gh_c.data_ptr<float>(), (const LevelNode*)nodes.data_ptr<int>(),
        hist_prefix.data_ptr<int>(), (const LevelWork*)work.data_ptr<int>(),
        (unsigned long long*)acc.data_ptr<int64_t>(), (int)k, (int)nfeat, (int)stride,
        (int)n_groups, (int)feats_per_group, gh_max.data_ptr<float>(), (int)rows_per_block,
        (int)slot_lo, (int)slot_hi);
  } else {
    launch_hist_device<short>(
        (int)grid, (int)hist_block, lds_bytes, stream, bins_c.data_ptr<short>(),
        (const float2*)gh_c.data_ptr<float>(), (const LevelNode*)nodes.data_ptr<int>(),
        hist_prefix.data_ptr<int>(), (const LevelWork*)work.data_ptr<int>(),
        (unsigned long long*)acc.data_ptr<int64_t>(), (int)k, (int)nfeat, (int)stride,
        (int)n_groups, (int)feats_per_group, gh_max.data_ptr<float>(), (int)rows_per_block,
        (int)slot_lo, (int)slot_hi);
  }
}

void grow_convert_level(torch::Tensor acc, torch::Tensor hist_f32, torch::Tensor nodes,
                        int64_t k, int64_t slots2, torch::Tensor gh_max) {
  const long long total = (long long)k * slots2;
  const int grid = (int)std::min<long long>((total + HIST_BLOCK - 1) / HIST_BLOCK, 2048);
  hipLaunchKernelGGL(convert_level_kernel, dim3(std::max(grid, 1)), dim3(HIST_BLOCK), 0,
                     current_stream(), (const unsigned long long*)acc.data_ptr<int64_t>(),
                     hist_f32.data_ptr<float>(), (const LevelNode*)nodes.data_ptr<int>(), (int)k,
                     slots2, gh_max.data_ptr<float>());
}

void grow_derive_level(torch::Tensor level_hist, torch::Tensor parent_hist, torch::Tensor nodes,
                       int64_t k, int64_t slots2) {
  const long long total = (long long)k * slots2;
  const int grid = (int)std::min<long long>((total + HIST_BLOCK - 1) / HIST_BLOCK, 2048);
  hipLaunchKernelGGL(derive_level_kernel, dim3(std::max(grid, 1)), dim3(HIST_BLOCK), 0,
                     current_stream(), level_hist.data_ptr<float>(),
                     parent_hist.data_ptr<float>(), (const LevelNode*)nodes.data_ptr<int>(),
                     (int)k, slots2);
}

// SMXGB_PART_LDS: 0 = classic two-pass partition, 1024/2048 (or "1" ->
// 1024) = LDS-staged single-source-pass variant (A/B experiment).
static int part_lds_mode() {
  static int mode = [] {
    const char* e = getenv("SMXGB_PART_LDS");
    if (e == nullptr) return 0;
    const int v = atoi(e);
    return v == 1 ? 1024 : v;
  }();
  return mode;
}

static bool launch_part_lds(int grid, hipStream_t stream, const unsigned char* src_bins,
                            const float2* src_gh, const int* src_rows,
                            unsigned char* dst_bins, float2* dst_gh, int* dst_rows,
                            const LevelNode* nodes, const int* part_prefix,
                            const LevelWork* work, const float* split_packed, int* counters,
                            int k, int nfeat, int missing_bin, int copy_payload) {
  const int tile = part_lds_mode();
  if (tile == 0 || (nfeat & 3) != 0) return false;
  const size_t lds = (size_t)tile * (8 + (size_t)(nfeat >> 2) * 4 + 8);
  if (lds > 158 * 1024) return false;
  if (tile == 2048) {
    hipLaunchKernelGGL(partition_device_lds_kernel<2048>, dim3(grid), dim3(HIST_BLOCK), lds,
                       stream, src_bins, src_gh, src_rows, dst_bins, dst_gh, dst_rows, nodes,
                       part_prefix, work, split_packed, counters, k, nfeat, missing_bin,
                       copy_payload);
  } else {
    hipLaunchKernelGGL(partition_device_lds_kernel<1024>, dim3(grid), dim3(HIST_BLOCK), lds,
                       stream, src_bins, src_gh, src_rows, dst_bins, dst_gh, dst_rows, nodes,
                       part_prefix, work, split_packed, counters, k, nfeat, missing_bin,
                       copy_payload);
  }
  return true;
}

void grow_partition_level(torch::Tensor src_bins, torch::Tensor src_gh, torch::Tensor src_rows,
                          torch::Tensor dst_bins, torch::Tensor dst_gh, torch::Tensor dst_rows,
                          torch::Tensor nodes, torch::Tensor part_prefix, torch::Tensor work,
                          torch::Tensor split_packed, torch::Tensor counters, int64_t k,
                          int64_t nfeat, int64_t missing_bin, int64_t grid,
                          int64_t copy_payload) {
  auto stream = current_stream();
  if (src_bins.scalar_type() == torch::kUInt8) {
    if (launch_part_lds((int)grid, stream, src_bins.data_ptr<unsigned char>(),
                        (const float2*)src_gh.data_ptr<float>(), src_rows.data_ptr<int>(),
                        dst_bins.data_ptr<unsigned char>(), (float2*)dst_gh.data_ptr<float>(),
                        dst_rows.data_ptr<int>(), (const LevelNode*)nodes.data_ptr<int>(),
                        part_prefix.data_ptr<int>(), (const LevelWork*)work.data_ptr<int>(),
                        split_packed.data_ptr<float>(), counters.data_ptr<int>(), (int)k,
                        (int)nfeat, (int)missing_bin, (int)copy_payload))
      return;
    hipLaunchKernelGGL(partition_device_kernel<unsigned char>, dim3((int)grid), dim3(HIST_BLOCK),
                       0, stream, src_bins.data_ptr<unsigned char>(),
                       (const float2*)src_gh.data_ptr<float>(), src_rows.data_ptr<int>(),
                       dst_bins.data_ptr<unsigned char>(), (float2*)dst_gh.data_ptr<float>(),
                       dst_rows.data_ptr<int>(), (const LevelNode*)nodes.data_ptr<int>(),
                       part_prefix.data_ptr<int>(), (const LevelWork*)work.data_ptr<int>(),
                       split_packed.data_ptr<float>(), counters.data_ptr<int>(), (int)k,
                       (int)nfeat, (int)missing_bin, (int)copy_payload);
  } else {
    hipLaunchKernelGGL(partition_device_kernel<short>, dim3((int)grid), dim3(HIST_BLOCK), 0,
                       stream, src_bins.data_ptr<short>(), (const float2*)src_gh.data_ptr<float>(),
                       src_rows.data_ptr<int>(), dst_bins.data_ptr<short>(),
                       (float2*)dst_gh.data_ptr<float>(), dst_rows.data_ptr<int>(),
                       (const LevelNode*)nodes.data_ptr<int>(), part_prefix.data_ptr<int>(),
                       (const LevelWork*)work.data_ptr<int>(), split_packed.data_ptr<float>(),
                       counters.data_ptr<int>(), (int)k, (int)nfeat, (int)missing_bin,
                       (int)copy_payload);
  }
}

// Whole-tree enqueue: the entire depthwise grow issued from C++ in one
// Python call (single-process path; the distributed path keeps the Python
// per-level loop so torch.distributed can enqueue the allreduce).
void grow_tree_enqueue(
    torch::Tensor init_bins, torch::Tensor init_gh, torch::Tensor init_rows,
    torch::Tensor bins0, torch::Tensor gh0, torch::Tensor rows0,
    torch::Tensor bins1, torch::Tensor gh1, torch::Tensor rows1,
    torch::Tensor nodes, torch::Tensor node_gh, torch::Tensor splits, torch::Tensor counts,
    torch::Tensor hist_f32, torch::Tensor acc, torch::Tensor cands,
    std::vector<torch::Tensor> hp, std::vector<torch::Tensor> pp, torch::Tensor work,
    torch::Tensor nbins, torch::Tensor feat_mask, torch::Tensor gh_max,
    int64_t D, int64_t cap, int64_t nfeat, int64_t stride, int64_t n_groups,
    int64_t feats_per_group, int64_t lds_words, int64_t has_missing, int64_t missing_bin,
    int64_t rows_per_block, int64_t max_blocks, int64_t hist_grid, int64_t part_grid,
    double reg_lambda, double reg_alpha, double gamma_, double min_child_weight,
    int64_t hist_block) {
  CHECK_GPU(init_bins);
  auto stream = current_stream();
  const long long slots2 = (long long)nfeat * stride * 2;
  const bool u8 = init_bins.scalar_type() == torch::kUInt8;
  const size_t lds_bytes = (size_t)lds_words * sizeof(unsigned long long);
  auto mono = torch::Tensor();  // device path excludes monotone constraints

  hipMemsetAsync(counts.data_ptr<int>(), 0, sizeof(int) * 2 * counts.size(0), stream);

  for (int d = 0; d < (int)D; ++d) {
    const int k = 1 << d;
    const int base = k - 1;
    LevelNode* nodes_d = (LevelNode*)nodes.data_ptr<int>() + base;
    float* gh_d = node_gh.data_ptr<float>() + (long long)base * 2;
    float* splits_d = splits.data_ptr<float>() + (long long)base * 6;
    int* counts_d = counts.data_ptr<int>() + (long long)base * 2;
    float* hist_d = hist_f32.data_ptr<float>() + (long long)base * slots2;
    int* hp_d = hp[d].data_ptr<int>();
    int* pp_d = pp[d].data_ptr<int>();
    LevelWork* work_d = (LevelWork*)(work.data_ptr<int>() + 2 * d);

    if (d == 0) {
      hipLaunchKernelGGL(make_root_kernel, dim3(1), dim3(1), 0, stream, nodes_d, hp_d, pp_d,
                         work_d, (int)cap, (int)rows_per_block, (int)max_blocks);
    } else {
      const int pk = k >> 1;
      const int pbase = pk - 1;
      hipLaunchKernelGGL(make_level_kernel, dim3(1), dim3(GROW_MAX_SLOTS), 0, stream,
                         (const LevelNode*)nodes.data_ptr<int>() + pbase,
                         splits.data_ptr<float>() + (long long)pbase * 6,
                         counts.data_ptr<int>() + (long long)pbase * 2,
                         node_gh.data_ptr<float>() + (long long)pbase * 2, nodes_d, gh_d, hp_d,
                         pp_d, work_d, pk, (int)rows_per_block, (int)max_blocks);
    }

    const bool level0 = d == 0;
    const bool par = (d % 2) == 1;
    const void* src_bins = level0 ? init_bins.data_ptr() : (par ? bins1.data_ptr() : bins0.data_ptr());
    const float2* src_gh = (const float2*)(level0 ? init_gh.data_ptr<float>()
                                                  : (par ? gh1.data_ptr<float>() : gh0.data_ptr<float>()));
    const int* src_rows = level0 ? init_rows.data_ptr<int>()
                                 : (par ? rows1.data_ptr<int>() : rows0.data_ptr<int>());
    const bool dpar = (d % 2) == 0;  // dst index = 1 - d%2
    void* dst_bins = dpar ? bins1.data_ptr() : bins0.data_ptr();
    float2* dst_gh = (float2*)(dpar ? gh1.data_ptr<float>() : gh0.data_ptr<float>());
    int* dst_rows = dpar ? rows1.data_ptr<int>() : rows0.data_ptr<int>();

    hipMemsetAsync(acc.data_ptr<int64_t>(), 0, sizeof(int64_t) * (size_t)k * slots2, stream);
    if (u8) {
      launch_hist_device<unsigned char>(
          (int)hist_grid, (int)hist_block, lds_bytes, stream, (const unsigned char*)src_bins,
          src_gh, nodes_d, hp_d, work_d, (unsigned long long*)acc.data_ptr<int64_t>(), k,
          (int)nfeat, (int)stride, (int)n_groups, (int)feats_per_group,
          gh_max.data_ptr<float>(), (int)rows_per_block);
    } else {
      launch_hist_device<short>(
          (int)hist_grid, (int)hist_block, lds_bytes, stream, (const short*)src_bins, src_gh,
          nodes_d, hp_d, work_d, (unsigned long long*)acc.data_ptr<int64_t>(), k, (int)nfeat,
          (int)stride, (int)n_groups, (int)feats_per_group, gh_max.data_ptr<float>(),
          (int)rows_per_block);
    }

    {
      const long long total = (long long)k * slots2;
      const int grid = (int)std::min<long long>((total + HIST_BLOCK - 1) / HIST_BLOCK, 2048);
      hipLaunchKernelGGL(convert_level_kernel, dim3(std::max(grid, 1)), dim3(HIST_BLOCK), 0,
                         stream, (const unsigned long long*)acc.data_ptr<int64_t>(), hist_d,
                         nodes_d, k, slots2, gh_max.data_ptr<float>());
      if (d > 0) {
        const int pbase = (k >> 1) - 1;
        hipLaunchKernelGGL(derive_level_kernel, dim3(std::max(grid, 1)), dim3(HIST_BLOCK), 0,
                           stream, hist_d, hist_f32.data_ptr<float>() + (long long)pbase * slots2,
                           nodes_d, k, slots2);
      }
    }

    const unsigned char* mask_ptr =
        feat_mask.numel() ? feat_mask.data_ptr<unsigned char>() : nullptr;
    hipLaunchKernelGGL(split_scan_kernel, dim3(k * (int)nfeat), dim3(SPLIT_BLOCK), 0, stream,
                       hist_d, (const float2*)gh_d, nbins.data_ptr<int>(), mask_ptr, nullptr,
                       (SplitCand*)cands.data_ptr<float>(), k, (int)nfeat, (int)stride,
                       (int)has_missing, 0, (float)reg_lambda, (float)reg_alpha, (float)gamma_,
                       (float)min_child_weight);
    hipLaunchKernelGGL(split_reduce_kernel, dim3(k), dim3(SPLIT_BLOCK), 0, stream,
                       (const SplitCand*)cands.data_ptr<float>(), splits_d, k, (int)nfeat);

    const int copy_payload = d < (int)D - 1 ? 1 : 0;  // last level: rows+counts only
    if (u8) {
      if (!launch_part_lds((int)part_grid, stream, (const unsigned char*)src_bins, src_gh,
                           src_rows, (unsigned char*)dst_bins, dst_gh, dst_rows, nodes_d, pp_d,
                           work_d, splits_d, counts_d, k, (int)nfeat, (int)missing_bin,
                           copy_payload))
        hipLaunchKernelGGL(partition_device_kernel<unsigned char>, dim3((int)part_grid),
                           dim3(HIST_BLOCK), 0, stream, (const unsigned char*)src_bins, src_gh,
                           src_rows, (unsigned char*)dst_bins, dst_gh, dst_rows, nodes_d, pp_d,
                           work_d, splits_d, counts_d, k, (int)nfeat, (int)missing_bin,
                           copy_payload);
    } else {
      hipLaunchKernelGGL(partition_device_kernel<short>, dim3((int)part_grid), dim3(HIST_BLOCK),
                         0, stream, (const short*)src_bins, src_gh, src_rows, (short*)dst_bins,
                         dst_gh, dst_rows, nodes_d, pp_d, work_d, splits_d, counts_d, k,
                         (int)nfeat, (int)missing_bin, copy_payload);
    }
  }
}

void grad_fused(torch::Tensor margin, torch::Tensor y, torch::Tensor w, torch::Tensor gh,
                torch::Tensor pmax, torch::Tensor psum, int64_t mode, double spw) {
  CHECK_GPU(margin);
  const long long n = margin.numel();
  const int grid = (int)pmax.size(0);
  hipLaunchKernelGGL(grad_fused_kernel, dim3(grid), dim3(HIST_BLOCK), 0, current_stream(),
                     margin.data_ptr<float>(), y.data_ptr<float>(),
                     w.numel() ? w.data_ptr<float>() : nullptr, (float2*)gh.data_ptr<float>(),
                     (float2*)pmax.data_ptr<float>(), (double2*)psum.data_ptr<double>(), n,
                     (int)mode, (float)spw);
}

void find_splits(torch::Tensor hist, torch::Tensor parent, torch::Tensor nbins,
                 torch::Tensor feat_mask, torch::Tensor monotone, torch::Tensor cands,
                 torch::Tensor out, int64_t k, int64_t f, int64_t stride, int64_t has_missing,
                 int64_t mask_per_node, double reg_lambda, double reg_alpha, double gamma_,
                 double min_child_weight) {
  CHECK_GPU(hist);
  auto stream = current_stream();
  const unsigned char* mask_ptr =
      feat_mask.numel() ? feat_mask.data_ptr<unsigned char>() : nullptr;
  const signed char* mono_ptr =
      monotone.numel() ? (const signed char*)monotone.data_ptr<int8_t>() : nullptr;
  hipLaunchKernelGGL(split_scan_kernel, dim3((int)(k * f)), dim3(SPLIT_BLOCK), 0, stream,
                     hist.data_ptr<float>(), (const float2*)parent.data_ptr<float>(),
                     nbins.data_ptr<int>(), mask_ptr, mono_ptr,
                     (SplitCand*)cands.data_ptr<float>(), (int)k, (int)f, (int)stride,
                     (int)has_missing, (int)mask_per_node, (float)reg_lambda, (float)reg_alpha,
                     (float)gamma_, (float)min_child_weight);
  hipLaunchKernelGGL(split_reduce_kernel, dim3((int)k), dim3(SPLIT_BLOCK), 0, stream,
                     (const SplitCand*)cands.data_ptr<float>(), out.data_ptr<float>(), (int)k,
                     (int)f);
}

void init_text_parsers(pybind11::module_& m);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  init_text_parsers(m);
  m.def("hist_build", &hist_build, "batched LDS-staged fixed-point histogram build");
  m.def("hist_convert", &hist_convert, "fixed-point -> float32 histogram convert");
  m.def("partition", &partition, "batched two-ended row partition");
  m.def("find_splits", &find_splits, "fused split-gain scan + per-node reduce");
  m.def("hist_build_compact", &hist_build_compact, "streaming histogram over compact row buffers");
  m.def("hist_convert_dev", &hist_convert_dev, "fixed-point -> f32 convert with device-resident scale");
  m.def("grow_make_root", &grow_make_root);
  m.def("grow_make_level", &grow_make_level);
  m.def("grow_hist_level", &grow_hist_level);
  m.def("grow_convert_level", &grow_convert_level);
  m.def("grow_derive_level", &grow_derive_level);
  m.def("grow_partition_level", &grow_partition_level);
  m.def("grow_tree_enqueue", &grow_tree_enqueue, "whole depthwise tree enqueued from one call");
  m.def("partition_compact", &partition_compact, "compacting partition (rows+bins+gh rewrite)");
  m.def("leaf_update_compact", &leaf_update_compact, "leaf scatter from compact row ids");
  m.def("leaf_update", &leaf_update, "batched leaf value scatter into margins");
  m.def("predict_forest", &predict_forest, "batched dense forest traversal");
  m.def("grad_fused", &grad_fused, "one-pass gradient + absmax partials");
  m.attr("_built_for") = "gfx950";
}
