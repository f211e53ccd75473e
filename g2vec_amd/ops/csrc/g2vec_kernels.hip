// g2vec_amd native kernels for AMD Instinct MI355X (gfx950, CDNA4).
//
// Hand-written HIP; no CUDA compatibility layers. Wavefront = 64 lanes
// throughout. Kernels (SURVEY 2.10 K-table in parentheses):
//
//   walk_kernel            (K10) CSR biased non-revisiting random walk,
//                          one wave per walk, LDS visited list, counter-based
//                          splitmix64 RNG (bit-identical to ops/cpu_ref.py)
//   cbow_fwd_scalar_kernel (K1+K2+K3+K8 collapsed) o_p = sum s_g over the
//                          path + fused sigmoid-CE loss/accuracy/dlogit
//   scatter_do_det_kernel  (K6 collapsed) c = X^T dO by gene-sorted segments
//                          — deterministic (no atomics)
//   adam_rank1_kernel      (K7) dense TF1-Adam with rank-1 grad c (x) who
//   adam_dense_kernel      (K7) dense TF1-Adam with materialized grad
//   cbow_fwd_kernel        (K1+K2+K3+K8 general) row-gather forward,
//                          bf16/f32 W_ih, fp32 accumulate, H stored
//   cbow_bwd_rows_det_kernel (K6 general) per-gene-segment dH-row reduce
//                          (deterministic, atomic-free)
//   pcc_edges_kernel       (K11) per-edge PCC dot product over samples
//   corr_gemm_kernel       (K11 dense path) C = Z Z^T / S on f32 MFMA
//                          (v_mfma_f32_16x16x4_f32) with LDS-staged tiles
//
// Reference behaviour being reimplemented (not copied): mathcom/G2Vec
// G2Vec.py:328-346 (walk), :238-251 (CBOW fwd/loss/acc), :245-246 (Adam),
// :354-368 (PCC).
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>
#include <cstdint>

// Debug build (G2VEC_DEBUG=1 at setup time): device-side bounds asserts in
// the index-following kernels (SURVEY 5.2 — the debug tier of the race/
// sanitizer story; host-side ASAN comes from the same flag's -fsanitize
// on the bindings TU).
#ifdef G2VEC_DEBUG
#define G2V_ASSERT(cond) \
  do { if (!(cond)) __builtin_trap(); } while (0)
#else
#define G2V_ASSERT(cond) do { } while (0)
#endif

#define WAVE 64

// ---------------------------------------------------------------- RNG / hash
// Must stay bit-identical to g2vec_amd/ops/cpu_ref.py::splitmix64/gene_hash.
__device__ __forceinline__ uint64_t sm64_next(uint64_t& s) {
  s += 0x9E3779B97F4A7C15ULL;
  uint64_t z = s;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}

__device__ __forceinline__ double u01_from(uint64_t r) {
  return (double)(r >> 11) * (1.0 / 9007199254740992.0);
}

__device__ __forceinline__ uint64_t gene_hash_dev(uint32_t g) {
  uint64_t s = (uint64_t)g * 0xBF58476D1CE4E5B9ULL + 0x9E3779B97F4A7C15ULL;
  return sm64_next(s);
}

// ---------------------------------------------------------------- wave utils
__device__ __forceinline__ int uni(int v) {
  return __builtin_amdgcn_readfirstlane(v);
}

__device__ __forceinline__ float unif(float v) {
  return __int_as_float(__builtin_amdgcn_readfirstlane(__float_as_int(v)));
}

__device__ __forceinline__ float wave_sum(float v) {
  for (int o = 32; o; o >>= 1) v += __shfl_down(v, o);
  return __shfl(v, 0);
}

// wave reduction with a PROVABLY-uniform (SGPR) result: downstream
// branches compile to scalar branches instead of exec-mask dances, and
// dependent integer math lands on the free scalar pipe
__device__ __forceinline__ float wave_sum_uni(float v) {
  for (int o = 32; o; o >>= 1) v += __shfl_down(v, o);
  return unif(v);   // lane 0 holds the total; full EXEC here
}

__device__ __forceinline__ float wave_incl_scan(float v) {
  const int lane = threadIdx.x & (WAVE - 1);
  for (int o = 1; o < WAVE; o <<= 1) {
    float t = __shfl_up(v, o);
    if (lane >= o) v += t;
  }
  return v;
}

__device__ __forceinline__ float bf16_to_f32(uint16_t b) {
  union { float f; uint32_t u; } x;
  x.u = ((uint32_t)b) << 16;
  return x.f;
}

// tag types so the gather kernel can distinguish the two 16-bit formats
struct bf16_bits { uint16_t v; };
struct fp16_bits { uint16_t v; };

__device__ __forceinline__ float to_f32(const bf16_bits b) {
  return bf16_to_f32(b.v);
}
__device__ __forceinline__ float to_f32(const fp16_bits h) {
  return __half2float(*(const __half*)&h.v);
}
__device__ __forceinline__ float to_f32(const float f) { return f; }

__device__ __forceinline__ uint16_t f32_to_bf16(float f) {
  union { float f; uint32_t u; } x;
  x.f = f;
  // round-to-nearest-even like torch's .bfloat16()
  uint32_t lsb = (x.u >> 16) & 1u;
  x.u += 0x7FFFu + lsb;
  return (uint16_t)(x.u >> 16);
}

// ================================================================= K10: walks
// One 64-lane wave per walk. Visited-set membership is an O(1) LDS
// open-addressing hash set (the naive O(path_len) visited-list scan per
// candidate left the CUs ~12% busy — latency-bound on the scan chain,
// measured via SQ_BUSY_CYCLES). For deg <= 64 the whole CSR row lives in
// one register per lane, so sampling is a single pass: masked weight ->
// wave sum -> wave prefix scan -> ballot pick. Matches the reference walk
// semantics exactly (G2Vec.py:328-346); the RNG stream is the framework's
// own (counter-based splitmix64, bit-shared with the CPU oracle).
#define HSET_EMPTY 0xFFFFFFFFu

__device__ __forceinline__ void hset_insert(uint32_t* tab, uint32_t tmask,
                                            uint32_t v) {
  uint32_t h = (v * 2654435761u) & tmask;
  for (;;) {
    const uint32_t c = tab[h];
    if (c == v) return;
    if (c == HSET_EMPTY) { tab[h] = v; return; }
    h = (h + 1) & tmask;
  }
}

__device__ __forceinline__ bool hset_contains(const uint32_t* tab,
                                              uint32_t tmask, uint32_t v) {
  uint32_t h = (v * 2654435761u) & tmask;
  for (;;) {
    const uint32_t c = tab[h];
    if (c == v) return true;
    if (c == HSET_EMPTY) return false;
    h = (h + 1) & tmask;
  }
}

// Rows with <= 4*64 neighbors sample entirely from registers (weights
// cached across the total/selection passes). Note: a packed int2
// {col,weight} layout was measured 19% SLOWER than separate col/weight
// arrays — CSR row starts have arbitrary parity, so half the dwordx2
// loads are 4-byte-misaligned.
#define WCHUNKS 4

extern "C" __global__ void __launch_bounds__(256)
walk_kernel(const int* __restrict__ row_ptr, const int* __restrict__ col_idx,
            const float* __restrict__ wgt, const int* __restrict__ sources,
            int n_src, long long n_walks, int num_rep, int len_path, int tsize,
            uint64_t seed, int* __restrict__ out_nodes,
            int* __restrict__ out_len, long long* __restrict__ out_hash) {
  extern __shared__ int smem[];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wib = uni(threadIdx.x >> 6);       // wave-in-block (SGPR)
  const int wpb = blockDim.x >> 6;             // waves per block
  int* vis = smem + wib * len_path;            // ordered path (output)
  uint32_t* tab = (uint32_t*)(smem + wpb * len_path) + wib * tsize;
  const uint32_t tmask = (uint32_t)tsize - 1;

  for (long long walk = (long long)blockIdx.x * wpb + wib; walk < n_walks;
       walk += (long long)gridDim.x * wpb) {
    const int rep = (int)(walk / n_src);
    const int src = uni(sources[walk % n_src]);
    // RNG keyed on the GLOBAL (source, repetition): DP-sharded generation
    // is bitwise-identical to single-process (C5 rank invariance)
    const uint64_t gid = (uint64_t)src * (uint64_t)num_rep + (uint64_t)rep;
    uint64_t state = seed ^ (uint64_t)(gid * 0x94D049BB133111EBULL + 1ULL);
    (void)sm64_next(state);                    // warm draw (CPU oracle parity)
    int cur = src;
    int plen = 0;
    uint64_t hash = 0;
    for (int i = lane; i < tsize; i += WAVE) tab[i] = HSET_EMPTY;

    for (int step = 0; step < len_path; ++step) {
      G2V_ASSERT(plen < len_path && cur >= 0);
      if (lane == 0) {
        vis[plen] = cur;
        // ONE lane inserts: a 64-lane same-address LDS store serializes its
        // lane group and the LDS array is shared by all 32 waves of the CU
        hset_insert(tab, tmask, (uint32_t)cur);
      }
      ++plen;
      hash += gene_hash_dev((uint32_t)cur);
      const int s = row_ptr[cur], e = row_ptr[cur + 1];
      const int deg = e - s;
      if (deg <= 0) break;

      if (deg <= WAVE) {
        // leanest path (the common case): one candidate per lane
        const int j = lane;
        int cand = -1;
        float w = 0.f;
        if (j < deg) {
          cand = col_idx[s + j];
          w = wgt[s + j];
          if (hset_contains(tab, tmask, (uint32_t)cand)) w = 0.f;
        }
        // one scan serves both the total (its lane-63 element) and the
        // selection — 6 fewer dependent ds_bpermute per step
        const float scan = wave_incl_scan(w);
        const float tot = unif(__shfl(scan, WAVE - 1));
        const uint64_t r = sm64_next(state);   // drawn even on dead end
        if (!(tot > 0.f)) break;
        const float target = (float)(u01_from(r) * (double)tot);
        const bool hit = (w > 0.f) && (scan > target) && (scan - w <= target);
        const unsigned long long mh = __ballot(hit);
        int lane_sel;
        if (mh != 0ULL) {
          lane_sel = __ffsll((long long)mh) - 1;
        } else {
          const unsigned long long mp = __ballot(w > 0.f);
          lane_sel = 63 - __clzll((long long)mp);
        }
        cur = uni(__shfl(cand, lane_sel));
      } else if (deg <= WCHUNKS * WAVE) {
        // register path: the whole row (<= 4 chunks of 64) is loaded once,
        // membership-masked once, and both the total and the selection use
        // the cached registers — one dwordx2 load + one hash probe per
        // candidate per STEP, nothing re-read.
        const int nchunk = (deg + WAVE - 1) >> 6;
        float wreg[WCHUNKS];
        int creg[WCHUNKS];
        float partial = 0.f;
#pragma unroll
        for (int k = 0; k < WCHUNKS; ++k) {
          wreg[k] = 0.f;
          creg[k] = -1;
          const int j = (k << 6) + lane;
          if (k < nchunk && j < deg) {
            const int cc = col_idx[s + j];
            creg[k] = cc;
            float w = wgt[s + j];
            if (hset_contains(tab, tmask, (uint32_t)cc)) w = 0.f;
            wreg[k] = w;
            partial += w;
          }
        }
        const float tot = wave_sum_uni(partial);
        const uint64_t r = sm64_next(state);   // drawn even on dead end
        if (!(tot > 0.f)) break;
        const float target = (float)(u01_from(r) * (double)tot);
        int chosen_cand = -1;
        float base = 0.f;
        unsigned long long any_pos = 0ULL;
        int last_pos_cand = -1;
        for (int k = 0; k < nchunk; ++k) {
          const float w = wreg[k];
          const float scan = wave_incl_scan(w);
          const float chunk_tot = unif(__shfl(scan, WAVE - 1));
          const bool hit = (w > 0.f) && (base + scan > target) &&
                           (base + scan - w <= target);
          const unsigned long long m = __ballot(hit);
          if (m != 0ULL) {
            chosen_cand = uni(__shfl(creg[k], __ffsll((long long)m) - 1));
            break;
          }
          const unsigned long long mp = __ballot(w > 0.f);
          if (mp != 0ULL) {
            any_pos = 1;
            last_pos_cand = uni(__shfl(creg[k], 63 - __clzll((long long)mp)));
          }
          base += chunk_tot;
        }
        if (chosen_cand < 0) {
          // rounding tail: target >= running total -> last unvisited
          if (!any_pos) break;                 // cannot happen when tot > 0
          chosen_cand = last_pos_cand;
        }
        cur = chosen_cand;
      } else {
        // chunked fallback for very-high-degree nodes (> 256 neighbors)
        float partial = 0.f;
        for (int j = lane; j < deg; j += WAVE) {
          const int cc = col_idx[s + j];
          float w = wgt[s + j];
          if (hset_contains(tab, tmask, (uint32_t)cc)) w = 0.f;
          partial += w;
        }
        const float tot = wave_sum_uni(partial);
        const uint64_t r = sm64_next(state);
        if (!(tot > 0.f)) break;
        const float target = (float)(u01_from(r) * (double)tot);
        int chosen = -1;
        float base = 0.f;
        for (int j0 = 0; j0 < deg; j0 += WAVE) {
          const int j = j0 + lane;
          float w = 0.f;
          if (j < deg) {
            const int cc = col_idx[s + j];
            w = wgt[s + j];
            if (hset_contains(tab, tmask, (uint32_t)cc)) w = 0.f;
          }
          const float scan = wave_incl_scan(w);
          const float chunk_tot = unif(__shfl(scan, WAVE - 1));
          const bool hit = (j < deg) && (w > 0.f) &&
                           (base + scan > target) && (base + scan - w <= target);
          const unsigned long long m = __ballot(hit);
          if (m != 0ULL) { chosen = j0 + (__ffsll((long long)m) - 1); break; }
          base += chunk_tot;
        }
        if (chosen < 0) {
          for (int j0 = ((deg - 1) / WAVE) * WAVE; j0 >= 0 && chosen < 0;
               j0 -= WAVE) {
            const int j = j0 + lane;
            float w = 0.f;
            if (j < deg) {
              const int cc = col_idx[s + j];
              w = wgt[s + j];
              if (hset_contains(tab, tmask, (uint32_t)cc)) w = 0.f;
            }
            const unsigned long long m = __ballot(w > 0.f);
            if (m != 0ULL) chosen = j0 + (63 - __clzll((long long)m));
          }
          if (chosen < 0) break;               // cannot happen when tot > 0
        }
        cur = uni(col_idx[s + chosen]);
      }
    }

    long long outb = walk * (long long)len_path;
    for (int k = lane; k < len_path; k += WAVE)
      out_nodes[outb + k] = (k < plen) ? vis[k] : -1;
    if (lane == 0) {
      out_len[walk] = plen;
      out_hash[walk] = (long long)hash;
    }
  }
}

// ===================================================== fast path: scalar CBOW
// o_p = sum_{g in p} s_g; loss = sigmoid-CE(o, y); correct = (o>0)==y;
// dO = (sigmoid(o)-y)*inv_b. One 16-LANE SUB-WAVE per path (4 paths per
// wavefront): paths average ~21 genes, so a full 64-lane wave left 2/3 of
// its lanes idle in both the gather and the reduce, and a thread-per-path
// variant serialized on gather latency (1.5x slower at 1M genes).
#define SUBW 16

__device__ __forceinline__ float subwave_sum16(float v) {
  v += __shfl_xor(v, 8);
  v += __shfl_xor(v, 4);
  v += __shfl_xor(v, 2);
  v += __shfl_xor(v, 1);
  return v;                       // every lane of the 16-group has the sum
}

extern "C" __global__ void __launch_bounds__(256)
cbow_fwd_scalar_kernel(const float* __restrict__ s, const int* __restrict__ genes,
                       const int* __restrict__ offs, const float* __restrict__ labels,
                       long long P, float inv_b, float* __restrict__ loss,
                       float* __restrict__ correct, float* __restrict__ dO) {
  const int sublane = threadIdx.x & (SUBW - 1);
  const int subs_per_block = blockDim.x / SUBW;
  const int sub = threadIdx.x / SUBW;
  for (long long p = (long long)blockIdx.x * subs_per_block + sub; p < P;
       p += (long long)gridDim.x * subs_per_block) {
    const int lo = offs[p], hi = offs[p + 1];
    float partial = 0.f;
    for (int i = lo + sublane; i < hi; i += SUBW) partial += s[genes[i]];
    const float o = subwave_sum16(partial);
    if (sublane == 0) {
      const float y = labels[p];
      loss[p] = fmaxf(o, 0.f) - o * y + log1pf(expf(-fabsf(o)));
      correct[p] = ((o > 0.f ? 1.f : 0.f) == y) ? 1.f : 0.f;
      if (dO) dO[p] = (1.f / (1.f + expf(-o)) - y) * inv_b;
    }
  }
}

// eval-only variant: nothing stored per path; the two splits' correct
// counts (train = paths < p_split of the concatenated set, val = rest)
// accumulate per wave, reduce in-block, and write ATOMIC-FREE per-block
// partials; a one-block second pass folds them (a same-two-words atomic
// fan-in measured ~11 ns per serialized add — 10k blocks cost 200 us)
extern "C" __global__ void __launch_bounds__(256)
cbow_eval_counts_kernel(const float* __restrict__ s, const int* __restrict__ genes,
                        const int* __restrict__ offs, const float* __restrict__ labels,
                        long long P, long long p_split,
                        float* __restrict__ partials,
                        float* __restrict__ dO, float inv_b) {
  const int sublane = threadIdx.x & (SUBW - 1);
  const int subs_per_block = blockDim.x / SUBW;
  const int sub = threadIdx.x / SUBW;
  float c0 = 0.f, c1 = 0.f;                    // sublane-0 accumulators
  for (long long p = (long long)blockIdx.x * subs_per_block + sub; p < P;
       p += (long long)gridDim.x * subs_per_block) {
    const int lo = offs[p], hi = offs[p + 1];
    float partial = 0.f;
    for (int i = lo + sublane; i < hi; i += SUBW)
      partial += s[__builtin_nontemporal_load(&genes[i])];
    const float o = subwave_sum16(partial);
    if (sublane == 0) {
      const float y = labels[p];
      const float corr = (((o > 0.f ? 1.f : 0.f) == y) ? 1.f : 0.f);
      if (p < p_split) c0 += corr; else c1 += corr;
      // fused next-epoch forward: this eval's s IS the next epoch's
      // pre-update s, so emit the train-split dO here and the standalone
      // forward kernel disappears from steady-state epochs (the s-gather
      // over the train paths runs ONCE per weight version, not twice)
      if (dO && p < p_split)
        dO[p] = (1.f / (1.f + expf(-o)) - y) * inv_b;
    }
  }
  // fold the per-thread accumulators: sublane0 lanes hold the values;
  // wave-reduce then block-reduce
  c0 = wave_sum(c0);
  c1 = wave_sum(c1);
  __shared__ float sm[8];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wib = threadIdx.x >> 6;
  if (lane == 0) { sm[wib] = c0; sm[4 + wib] = c1; }
  __syncthreads();
  if (threadIdx.x == 0) {
    partials[2 * blockIdx.x] = sm[0] + sm[1] + sm[2] + sm[3];
    partials[2 * blockIdx.x + 1] = sm[4] + sm[5] + sm[6] + sm[7];
  }
}

// LDS-staged variant: at ex_* scale the whole s table is 30 KB, but the
// subwave gathers above are TA-throughput bound — a 64-lane gather over a
// 30 KB range touches ~40-60 distinct L1 lines, serialized in the CU's
// one texture/addressing unit. Staging s in LDS turns each gather into a
// ds_read (32-bank parallel, ~2-4x conflict factor), so the block pays
// one coalesced G-float stage (L2-broadcast: every block reads the same
// table) and then gathers at LDS rate. The per-path math and accumulation
// order are IDENTICAL to cbow_eval_counts_kernel — bitwise-equal outputs;
// the host picks this kernel when G*4 fits the LDS budget and caps the
// grid so each resident block amortizes its stage over many paths.
extern "C" __global__ void __launch_bounds__(256)
cbow_eval_counts_lds_kernel(const float* __restrict__ s,
                            const int* __restrict__ genes,
                            const int* __restrict__ offs,
                            const float* __restrict__ labels,
                            long long P, long long p_split,
                            float* __restrict__ partials,
                            float* __restrict__ dO, float inv_b, int G) {
  extern __shared__ float s_lds[];
  for (int g = threadIdx.x; g < G; g += blockDim.x) s_lds[g] = s[g];
  __syncthreads();
  const int sublane = threadIdx.x & (SUBW - 1);
  const int subs_per_block = blockDim.x / SUBW;
  const int sub = threadIdx.x / SUBW;
  float c0 = 0.f, c1 = 0.f;
  for (long long p = (long long)blockIdx.x * subs_per_block + sub; p < P;
       p += (long long)gridDim.x * subs_per_block) {
    const int lo = offs[p], hi = offs[p + 1];
    float partial = 0.f;
    for (int i = lo + sublane; i < hi; i += SUBW)
      partial += s_lds[__builtin_nontemporal_load(&genes[i])];
    const float o = subwave_sum16(partial);
    if (sublane == 0) {
      const float y = labels[p];
      const float corr = (((o > 0.f ? 1.f : 0.f) == y) ? 1.f : 0.f);
      if (p < p_split) c0 += corr; else c1 += corr;
      if (dO && p < p_split)
        dO[p] = (1.f / (1.f + expf(-o)) - y) * inv_b;
    }
  }
  c0 = wave_sum(c0);
  c1 = wave_sum(c1);
  __shared__ float sm[8];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wib = threadIdx.x >> 6;
  if (lane == 0) { sm[wib] = c0; sm[4 + wib] = c1; }
  __syncthreads();
  if (threadIdx.x == 0) {
    partials[2 * blockIdx.x] = sm[0] + sm[1] + sm[2] + sm[3];
    partials[2 * blockIdx.x + 1] = sm[4] + sm[5] + sm[6] + sm[7];
  }
}

extern "C" __global__ void __launch_bounds__(256)
fold_partials_kernel(const float* __restrict__ partials, int n_blocks,
                     float* __restrict__ counts) {
  float c0 = 0.f, c1 = 0.f;
  for (int b = threadIdx.x; b < n_blocks; b += blockDim.x) {
    c0 += partials[2 * b];
    c1 += partials[2 * b + 1];
  }
  c0 = wave_sum(c0);
  c1 = wave_sum(c1);
  __shared__ float sm[8];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wib = threadIdx.x >> 6;
  if (lane == 0) { sm[wib] = c0; sm[4 + wib] = c1; }
  __syncthreads();
  if (threadIdx.x == 0) {
    counts[0] = sm[0] + sm[1] + sm[2] + sm[3];
    counts[1] = sm[4] + sm[5] + sm[6] + sm[7];
  }
}

// =============================================== fast path: c = X^T dO (det.)
// One wave per gene segment (instances pre-sorted by gene, plan built once
// per path set). Fixed tree-reduction order -> bitwise reproducible.
extern "C" __global__ void __launch_bounds__(256)
scatter_do_det_kernel(const int* __restrict__ inst_path,
                      const int* __restrict__ seg_start,
                      const int* __restrict__ seg_gene, int n_seg,
                      const float* __restrict__ dO, float* __restrict__ c) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wib = threadIdx.x >> 6;
  const int wpb = blockDim.x >> 6;
  for (long long sidx = (long long)blockIdx.x * wpb + wib; sidx < n_seg;
       sidx += (long long)gridDim.x * wpb) {
    const int lo = seg_start[sidx], hi = seg_start[sidx + 1];
    float p0 = 0.f, p1 = 0.f, p2 = 0.f, p3 = 0.f;
    int i = lo + lane;
    for (; i + 3 * WAVE < hi; i += 4 * WAVE) {
      p0 += dO[inst_path[i]];
      p1 += dO[inst_path[i + WAVE]];
      p2 += dO[inst_path[i + 2 * WAVE]];
      p3 += dO[inst_path[i + 3 * WAVE]];
    }
    for (; i < hi; i += WAVE) p0 += dO[inst_path[i]];
    const float v = wave_sum((p0 + p1) + (p2 + p3));
    if (lane == 0) c[seg_gene[sidx]] += v;   // += so slab-blocked passes
                                             // accumulate (c starts zeroed;
                                             // launches are stream-ordered,
                                             // so still deterministic)
  }
}

// ============================================================== K7: TF1 Adam
// theta -= lr_t * m / (sqrt(v) + eps); lr_t precomputed on host.
// rank-1 variant forms grad[g*h+j] = c[g] * who[j] on the fly (the linear
// CBOW's dW_ih is rank-1, see models/cbow.py) — no G*h grad tensor exists.
typedef float f32x4 __attribute__((ext_vector_type(4)));

extern "C" __global__ void __launch_bounds__(256)
adam_rank1_kernel(float* __restrict__ W, float* __restrict__ m,
                  float* __restrict__ v, const float* __restrict__ c,
                  const float* __restrict__ who, long long G, int h,
                  const float* __restrict__ lr_t_ptr, float b1, float b2,
                  float eps, float* __restrict__ gw_partials) {
  // gw_partials != nullptr: ALSO emit this block's partial of
  // dW_ho = W_pre^T c ([gridDim.x, h] layout) — the fused epilogue reads
  // the pre-update W rows this kernel loads anyway, removing the
  // separate gemv_cols pass (a full W read) and its fold launch.
  // Deterministic: each thread accumulates its own grid-stride rows in
  // order, then fixed ascending-lrow LDS tree per block, then the fold
  // kernel sums blocks ascending.
  const float lr_t = lr_t_ptr[0];   // device-read so hipGraph replays see
                                    // the per-step bias-corrected value
  f32x4* W4 = (f32x4*)W; f32x4* m4 = (f32x4*)m; f32x4* v4 = (f32x4*)v;
  const f32x4* who4 = (const f32x4*)who;
  const int h4 = h / 4;
  // row-major indexing: the flat-index variant paid a 64-bit div+mod per
  // f32x4; here each thread owns a fixed column group and strides over
  // rows (one div/mod per THREAD at setup, multiply-add per iteration)
  const int rpb = blockDim.x / h4;             // rows per block (h <= 1024)
  const int lrow = (int)threadIdx.x / h4;      // this thread's row-in-block
  const int j4 = (int)threadIdx.x - lrow * h4; // and column group
  const f32x4 wj = who4[j4];
  f32x4 gw_acc = {0.f, 0.f, 0.f, 0.f};
  for (long long g = (long long)blockIdx.x * rpb + lrow; g < G;
       g += (long long)gridDim.x * rpb) {
    const long long i = g * h4 + j4;
    const float cg = c[g];
    // W/m/v are pure streams (no reuse inside an epoch): nontemporal
    // ld/st keeps them from evicting the gather tables in the XCD L2s
    f32x4 mm = __builtin_nontemporal_load(&m4[i]);
    f32x4 vv = __builtin_nontemporal_load(&v4[i]);
    f32x4 ww = __builtin_nontemporal_load(&W4[i]);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      if (gw_partials) gw_acc[k] += cg * ww[k];   // pre-update W row
      const float grad = cg * wj[k];
      mm[k] = b1 * mm[k] + (1.f - b1) * grad;
      vv[k] = b2 * vv[k] + (1.f - b2) * grad * grad;
      ww[k] -= lr_t * mm[k] / (sqrtf(vv[k]) + eps);
    }
    __builtin_nontemporal_store(mm, &m4[i]);
    __builtin_nontemporal_store(vv, &v4[i]);
    __builtin_nontemporal_store(ww, &W4[i]);
  }
  if (gw_partials) {
    // per-block fold: slots [rpb][h] in LDS, summed ascending lrow
    __shared__ float gw_slots[256 * 4];          // rpb * h = 1024 floats
    float* slot = gw_slots + (lrow * h4 + j4) * 4;
#pragma unroll
    for (int k = 0; k < 4; ++k) slot[k] = gw_acc[k];
    __syncthreads();
    if (lrow == 0) {
      f32x4 tot = {0.f, 0.f, 0.f, 0.f};
      for (int r = 0; r < rpb; ++r) {
        const float* sl = gw_slots + (r * h4 + j4) * 4;
#pragma unroll
        for (int k = 0; k < 4; ++k) tot[k] += sl[k];
      }
      float* out = gw_partials + (long long)blockIdx.x * h + j4 * 4;
#pragma unroll
      for (int k = 0; k < 4; ++k) out[k] = tot[k];
    }
  }
}

// fold the adam_rank1 gw partials and apply the TF1-Adam update to W_ho
// in the same launch. One WAVE per column: lanes stride the block
// partials (fixed stride + fixed shuffle tree -> deterministic); the
// single-block variant was latency-bound at ~1k dependent loads/column.
extern "C" __global__ void __launch_bounds__(256)
fold_gw_adam_kernel(const float* __restrict__ partials, int n_blocks, int h,
                    float* __restrict__ who, float* __restrict__ mO,
                    float* __restrict__ vO,
                    const float* __restrict__ lr_t_ptr, float b1, float b2,
                    float eps) {
  const float lr_t = lr_t_ptr[0];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wib = threadIdx.x >> 6;
  const int wpb = blockDim.x >> 6;
  for (int i = blockIdx.x * wpb + wib; i < h; i += gridDim.x * wpb) {
    float g = 0.f;
    for (int b = lane; b < n_blocks; b += WAVE)
      g += partials[(long long)b * h + i];
    g = wave_sum(g);
    if (lane == 0) {
      const float mm = b1 * mO[i] + (1.f - b1) * g;
      const float vv = b2 * vO[i] + (1.f - b2) * g * g;
      mO[i] = mm;
      vO[i] = vv;
      who[i] -= lr_t * mm / (sqrtf(vv) + eps);
    }
  }
}

extern "C" __global__ void __launch_bounds__(256)
adam_dense_kernel(float* __restrict__ W, float* __restrict__ m,
                  float* __restrict__ v, const float* __restrict__ grad,
                  long long n, const float* __restrict__ lr_t_ptr, float b1,
                  float b2, float eps) {
  const float lr_t = lr_t_ptr[0];
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long long)gridDim.x * blockDim.x) {
    const float g = grad[i];
    float mm = b1 * m[i] + (1.f - b1) * g;
    float vv = b2 * v[i] + (1.f - b2) * g * g;
    m[i] = mm; v[i] = vv;
    W[i] -= lr_t * mm / (sqrtf(vv) + eps);
  }
}

// ====================================== general path: row-gather CBOW forward
// One wave per path; h/64 embedding columns per lane (contiguous, vector
// loads); f32 accumulate; H stored for the general backward; loss/acc/dO
// fused into the same pass (the reference recomputed the forward 3x per
// epoch and shipped 2.5 GB/epoch host->device, SURVEY 3.2 — here everything
// is resident and fused).
template <typename WT, int HPL>
__global__ void __launch_bounds__(256)
cbow_fwd_kernel(const WT* __restrict__ W, const float* __restrict__ who,
                const int* __restrict__ genes, const int* __restrict__ offs,
                const float* __restrict__ labels, long long P, float inv_b,
                int h, float* __restrict__ H, float* __restrict__ loss,
                float* __restrict__ correct, float* __restrict__ dO,
                int act) {
  // act: 0 = linear (reference semantics, G2Vec.py:238-240); 1 = ReLU on
  // the hidden vector (opt-in non-linear successor; H stores the
  // PRE-activation so the backward recovers the mask)
  const int lane = threadIdx.x & (WAVE - 1);
  const int wib = threadIdx.x >> 6;
  const int wpb = blockDim.x >> 6;
  const int col0 = lane * HPL;
  for (long long p = (long long)blockIdx.x * wpb + wib; p < P;
       p += (long long)gridDim.x * wpb) {
    const int lo = offs[p], hi = offs[p + 1];
    float acc[HPL];
#pragma unroll
    for (int k = 0; k < HPL; ++k) acc[k] = 0.f;
    for (int i = lo; i < hi; ++i) {
      const long long g = genes[i];
      G2V_ASSERT(g >= 0);
      const WT* row = W + g * (long long)h + col0;
      WT tmp[HPL];
      __builtin_memcpy(tmp, row, HPL * sizeof(WT));  // one wide load
#pragma unroll
      for (int k = 0; k < HPL; ++k) acc[k] += to_f32(tmp[k]);
    }
    float op = 0.f;
    float wv[HPL];
    __builtin_memcpy(wv, who + col0, HPL * sizeof(float));
#pragma unroll
    for (int k = 0; k < HPL; ++k)
      op += (act ? fmaxf(acc[k], 0.f) : acc[k]) * wv[k];
    const float o = wave_sum(op);
    if (H) {
      float* hrow = H + p * (long long)h + col0;
      __builtin_memcpy(hrow, acc, HPL * sizeof(float));
    }
    if (lane == 0) {
      const float y = labels[p];
      loss[p] = fmaxf(o, 0.f) - o * y + log1pf(expf(-fabsf(o)));
      correct[p] = ((o > 0.f ? 1.f : 0.f) == y) ? 1.f : 0.f;
      if (dO) dO[p] = (1.f / (1.f + expf(-o)) - y) * inv_b;
    }
  }
}

// ======================================= general path: dW_ih scatter backward
// Deterministic, atomic-free: instances are pre-sorted by gene (the same
// ScatterPlan as the fast path's c-reduction); one wave per gene segment
// accumulates its instances' dH rows in registers and writes the dW row
// once. dH_p = dO_p * who is recomputed per instance — for a non-linear
// successor, swap that line for a load from a materialized dH[P, h].
// (The v1 atomicAdd-per-touched-row kernel spent 2.1 ms/epoch on hub-gene
// contention at ex_* scale; this is ~100x less.)
template <int HPL>
__global__ void __launch_bounds__(256)
cbow_bwd_rows_det_kernel(const float* __restrict__ who,
                         const int* __restrict__ inst_path,
                         const int* __restrict__ seg_start,
                         const int* __restrict__ seg_gene, int n_seg,
                         const float* __restrict__ dO, int h,
                         float* __restrict__ dW,
                         const float* __restrict__ Hpre) {
  // Hpre != nullptr: ReLU backward — dH_p = dO_p * who (.) 1[Hpre_p > 0]
  // (one extra wide load per instance; linear path passes nullptr)
  const int lane = threadIdx.x & (WAVE - 1);
  const int wib = threadIdx.x >> 6;
  const int wpb = blockDim.x >> 6;
  const int col0 = lane * HPL;
  float wv[HPL];
  __builtin_memcpy(wv, who + col0, HPL * sizeof(float));
  for (long long s = (long long)blockIdx.x * wpb + wib; s < n_seg;
       s += (long long)gridDim.x * wpb) {
    const int lo = seg_start[s], hi = seg_start[s + 1];
    float acc[HPL];
#pragma unroll
    for (int k = 0; k < HPL; ++k) acc[k] = 0.f;
    if (Hpre) {
      for (int i = lo; i < hi; ++i) {
        const long long pi = inst_path[i];
        const float g = dO[pi];
        float hrow[HPL];
        __builtin_memcpy(hrow, Hpre + pi * (long long)h + col0,
                         HPL * sizeof(float));
#pragma unroll
        for (int k = 0; k < HPL; ++k)
          acc[k] += (hrow[k] > 0.f) ? g * wv[k] : 0.f;
      }
    } else {
      for (int i = lo; i < hi; ++i) {
        const float g = dO[inst_path[i]];     // general shape: dH row source
#pragma unroll
        for (int k = 0; k < HPL; ++k) acc[k] += g * wv[k];
      }
    }
    float* row = dW + (long long)seg_gene[s] * h + col0;
    __builtin_memcpy(row, acc, HPL * sizeof(float));
  }
}

// =================================================== K11: per-edge PCC weight
// zt: f32 [G, S] z-scored rows; one wave per edge, lane-strided dot.
// PCC = dot / S; 0-variance genes have zero rows (graph.py) -> pcc 0.
extern "C" __global__ void __launch_bounds__(256)
pcc_edges_kernel(const float* __restrict__ zt, const int* __restrict__ edges,
                 long long E, int S, float inv_s, float* __restrict__ out) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wib = threadIdx.x >> 6;
  const int wpb = blockDim.x >> 6;
  for (long long e = (long long)blockIdx.x * wpb + wib; e < E;
       e += (long long)gridDim.x * wpb) {
    const long long a = edges[2 * e], b = edges[2 * e + 1];
    const float* za = zt + a * S;
    const float* zb = zt + b * S;
    float partial = 0.f;
    for (int i = lane; i < S; i += WAVE) partial += za[i] * zb[i];
    const float d = wave_sum(partial);
    if (lane == 0) out[e] = fabsf(d * inv_s);
  }
}

// ============================================== K11 dense: MFMA f32 corr GEMM
// C = Zt Zt^T / S on v_mfma_f32_16x16x4_f32 (exact f32 at the FP32 matrix
// rate — no xf32 on gfx950, see cdna_hip_programming.md §3). 64x64 tile per
// 4-wave block (each wave one 32x32 sub-tile = 2x2 16x16 fragments), both
// operand tiles staged once in LDS (K = S <= 288 fits the 160 KiB budget).
typedef float f32x4c __attribute__((ext_vector_type(4)));

extern "C" __global__ void __launch_bounds__(256)
corr_gemm_kernel(const float* __restrict__ zt, float* __restrict__ C,
                 int G, int S, int S4, float inv_s) {
#if defined(__gfx950__)
  extern __shared__ float lds[];
  float* At = lds;                 // [64][S4]
  float* Bt = lds + 64 * S4;       // [64][S4]
  const int ntile = (G + 63) / 64;
  const int ti = blockIdx.x / ntile;
  const int tj = blockIdx.x % ntile;
  const int i0 = ti * 64, j0 = tj * 64;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x >> 6;
  // stage: each thread strides over the 64 x S4 tiles
  for (int idx = threadIdx.x; idx < 64 * S4; idx += blockDim.x) {
    const int r = idx / S4, k = idx % S4;
    At[idx] = (i0 + r < G && k < S) ? zt[(long long)(i0 + r) * S + k] : 0.f;
    Bt[idx] = (j0 + r < G && k < S) ? zt[(long long)(j0 + r) * S + k] : 0.f;
  }
  __syncthreads();

  const int wr = wid >> 1, wc = wid & 1;       // 2x2 wave grid -> 32x32 each
  f32x4c acc[2][2];
#pragma unroll
  for (int a = 0; a < 2; ++a)
#pragma unroll
    for (int b = 0; b < 2; ++b) acc[a][b] = (f32x4c)(0.f);

  const int fr = lane & 15;                     // fragment row/col lane index
  const int kk = lane >> 4;                     // k sub-index 0..3
  for (int k0 = 0; k0 < S4; k0 += 4) {
    float av[2], bv[2];
#pragma unroll
    for (int a = 0; a < 2; ++a)
      av[a] = At[(wr * 32 + a * 16 + fr) * S4 + k0 + kk];
#pragma unroll
    for (int b = 0; b < 2; ++b)
      bv[b] = Bt[(wc * 32 + b * 16 + fr) * S4 + k0 + kk];
#pragma unroll
    for (int a = 0; a < 2; ++a)
#pragma unroll
      for (int b = 0; b < 2; ++b)
        acc[a][b] = __builtin_amdgcn_mfma_f32_16x16x4f32(av[a], bv[b],
                                                         acc[a][b], 0, 0, 0);
  }
  // C/D map (16x16): col = lane&15, row = (lane>>4)*4 + reg
#pragma unroll
  for (int a = 0; a < 2; ++a)
#pragma unroll
    for (int b = 0; b < 2; ++b)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int ci = i0 + wr * 32 + a * 16 + (lane >> 4) * 4 + reg;
        const int cj = j0 + wc * 32 + b * 16 + (lane & 15);
        if (ci < G && cj < G)
          C[(long long)ci * G + cj] = acc[a][b][reg] * inv_s;
      }
#endif
}

// ============================================== hot-path GEMVs (own kernels)
// rocBLAS gemvn on the tall-skinny [G, h] x [h] shape ran at ~2% of HBM
// bandwidth (18.6 ms for the 1M x 512 s = W_ih @ W_ho). One wave per row,
// h/64 contiguous columns per lane, wave-reduce: HBM-roofline instead.
template <int HPL>
__global__ void __launch_bounds__(256)
gemv_rows_kernel(const float* __restrict__ W, const float* __restrict__ x,
                 long long G, int h, float* __restrict__ out) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wib = threadIdx.x >> 6;
  const int wpb = blockDim.x >> 6;
  const int col0 = lane * HPL;
  float xv[HPL];
  __builtin_memcpy(xv, x + col0, HPL * sizeof(float));
  for (long long g = (long long)blockIdx.x * wpb + wib; g < G;
       g += (long long)gridDim.x * wpb) {
    float acc[HPL];
    __builtin_memcpy(acc, W + g * (long long)h + col0, HPL * sizeof(float));
    float p = 0.f;
#pragma unroll
    for (int k = 0; k < HPL; ++k) p += acc[k] * xv[k];
    const float o = wave_sum(p);
    if (lane == 0) out[g] = o;
  }
}

// out[j] = sum_g c[g] * W[g, j] (the dW_ho = W_ih^T c reduction): each
// block accumulates a row range into a private [h] partial; a second pass
// folds the partials (atomic-free).
template <int CPT>   // columns per thread = max(1, h/256)
__global__ void __launch_bounds__(256)
gemv_cols_kernel(const float* __restrict__ W, const float* __restrict__ c,
                 long long G, int h, float* __restrict__ partials) {
  const int col0 = threadIdx.x * CPT;
  if (col0 >= h) return;            // h < 256: surplus threads idle
  float acc[CPT];
#pragma unroll
  for (int k = 0; k < CPT; ++k) acc[k] = 0.f;
  const long long rows_per_blk = (G + gridDim.x - 1) / gridDim.x;
  const long long lo = (long long)blockIdx.x * rows_per_blk;
  const long long hi = (lo + rows_per_blk < G) ? lo + rows_per_blk : G;
  for (long long g = lo; g < hi; ++g) {
    const float cg = c[g];
    float wv[CPT];
    __builtin_memcpy(wv, W + g * (long long)h + col0, CPT * sizeof(float));
#pragma unroll
    for (int k = 0; k < CPT; ++k) acc[k] += cg * wv[k];
  }
  float* p = partials + (long long)blockIdx.x * h + col0;
  __builtin_memcpy(p, acc, CPT * sizeof(float));
}

extern "C" __global__ void __launch_bounds__(256)
fold_cols_kernel(const float* __restrict__ partials, int n_blocks, int h,
                 float* __restrict__ out) {
  // one wave per column, lane-strided over the block partials (a serial
  // per-thread loop over ~1k uncoalesced partial rows measured ~150 us)
  const int lane = threadIdx.x & (WAVE - 1);
  const int col = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (col >= h) return;
  float a = 0.f;
  for (int b = lane; b < n_blocks; b += WAVE)
    a += partials[(long long)b * h + col];
  a = wave_sum(a);
  if (lane == 0) out[col] = a;
}

// =============================================================== bf16 cast
extern "C" __global__ void __launch_bounds__(256)
f32_to_bf16_kernel(const float* __restrict__ in, uint16_t* __restrict__ out,
                   long long n) {
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long long)gridDim.x * blockDim.x)
    out[i] = f32_to_bf16(in[i]);
}

// ------------------------------------------------- template instantiations
#define INSTANTIATE_FWD(WT, HPL)                                              \
  template __global__ void cbow_fwd_kernel<WT, HPL>(                          \
      const WT*, const float*, const int*, const int*, const float*,          \
      long long, float, int, float*, float*, float*, float*, int);
INSTANTIATE_FWD(float, 1)
INSTANTIATE_FWD(float, 2)
INSTANTIATE_FWD(float, 4)
INSTANTIATE_FWD(float, 8)
INSTANTIATE_FWD(float, 16)
INSTANTIATE_FWD(bf16_bits, 1)
INSTANTIATE_FWD(bf16_bits, 2)
INSTANTIATE_FWD(bf16_bits, 4)
INSTANTIATE_FWD(bf16_bits, 8)
INSTANTIATE_FWD(bf16_bits, 16)
INSTANTIATE_FWD(fp16_bits, 1)
INSTANTIATE_FWD(fp16_bits, 2)
INSTANTIATE_FWD(fp16_bits, 4)
INSTANTIATE_FWD(fp16_bits, 8)
INSTANTIATE_FWD(fp16_bits, 16)

#define INSTANTIATE_BWD(HPL)                                                  \
  template __global__ void cbow_bwd_rows_det_kernel<HPL>(                     \
      const float*, const int*, const int*, const int*, int, const float*,    \
      int, float*, const float*);
INSTANTIATE_BWD(1)
INSTANTIATE_BWD(2)
INSTANTIATE_BWD(4)
INSTANTIATE_BWD(8)
INSTANTIATE_BWD(16)

#define INSTANTIATE_GEMVR(HPL)                                                \
  template __global__ void gemv_rows_kernel<HPL>(                             \
      const float*, const float*, long long, int, float*);
INSTANTIATE_GEMVR(1)
INSTANTIATE_GEMVR(2)
INSTANTIATE_GEMVR(4)
INSTANTIATE_GEMVR(8)
INSTANTIATE_GEMVR(16)

#define INSTANTIATE_GEMVC(CPT)                                                \
  template __global__ void gemv_cols_kernel<CPT>(                             \
      const float*, const float*, long long, int, float*);
INSTANTIATE_GEMVC(1)
INSTANTIATE_GEMVC(2)
INSTANTIATE_GEMVC(4)

// ======================================================= K9: trunc-normal init
// Seeded device truncated normal (+-2 sigma, rejection-resampled) — the
// reference init semantics (tf.truncated_normal, G2Vec.py:234-235)
// computed entirely on-device: the host rejection loop materialized
// multi-GB intermediates at the 1M x 512 config (round-1 verdict item 6).
// Counter-based: element i draws from its own splitmix64 stream, so the
// result is independent of the launch geometry and bit-stable across
// runs; the CPU oracle (cpu_ref.trunc_normal) mirrors the algorithm.
extern "C" __global__ void trunc_normal_kernel(float* __restrict__ out,
                                               long long n, float std,
                                               uint64_t seed) {
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += stride) {
    uint64_t state =
        seed ^ (uint64_t)((uint64_t)i * 0x94D049BB133111EBULL + 1ULL);
    (void)sm64_next(state);                  // warm draw (stream decorrelation)
    float x;
    do {                                     // P(reject) ~ 4.6%/draw
      const double u1 = u01_from(sm64_next(state));
      const double u2 = u01_from(sm64_next(state));
      // Box-Muller (cosine branch); u1 clamped away from log(0)
      const double r = sqrt(-2.0 * log(u1 > 1e-300 ? u1 : 1e-300));
      x = (float)(r * cos(6.283185307179586 * u2));
    } while (fabsf(x) > 2.0f);
    out[i] = x * std;
  }
}

// ================================= fast path: instance-parallel fused eval
// The subwave-per-path eval kernel is latency-bound: ~21-gene paths give
// each 16-lane group a 2-iteration dependent gather chain and idle lanes
// (26 us over 1.5M instances at ex_* scale). This pair streams the flat
// CSR instance space instead: every lane gathers ONE s value, a 64-lane
// segmented inclusive scan (head flags at path boundaries) produces
// per-path piece sums per 64-instance window, and a thread-per-path
// finish kernel folds each path's <= cap pieces (fixed ascending window
// order -> deterministic), computes correctness counts for the two
// splits, and emits the train split's next-epoch dlogits (same fusion
// as cbow_eval_counts_kernel). Wave windows are aligned: every wave
// covers instances [w*64, w*64+64).
extern "C" __global__ void __launch_bounds__(256)
eval_scan_kernel(const float* __restrict__ s, const int* __restrict__ genes,
                 const int* __restrict__ pathid, const int* __restrict__ offs,
                 long long nnz, int cap, float* __restrict__ piece,
                 int g_lds) {
  // g_lds > 0: the whole s table is staged into LDS once per
  // (persistent, grid-strided) block — random 4-byte s-gathers become
  // ds_read ops instead of 64-way-divergent L1 line lookups, the
  // measured wall of the gather at <= 48 KB tables
  extern __shared__ float s_lds[];
  const float* st = s;
  if (g_lds > 0) {
    for (int g = threadIdx.x; g < g_lds; g += blockDim.x) s_lds[g] = s[g];
    __syncthreads();
    st = s_lds;
  }
  const int lane = threadIdx.x & (WAVE - 1);
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i - lane < nnz; i += stride) {           // whole wave iterates together
    const bool act = i < nnz;
    const int pid = act ? pathid[i] : -1;
    float val = act ? st[genes[i]] : 0.f;
    const int pid_up = __shfl_up(pid, 1);
    int f = (lane == 0) || (pid != pid_up);     // head of a window piece
    const int head = f;
#pragma unroll
    for (int o = 1; o < WAVE; o <<= 1) {        // segmented inclusive scan
      const float tv = __shfl_up(val, o);
      const int tf = __shfl_up(f, o);
      if (lane >= o) {
        if (!f) val += tv;
        f = f | tf;
      }
    }
    const int head_down = __shfl_down(head, 1);
    const bool last = (lane == WAVE - 1) || head_down;
    if (act && last) {
      const long long w = i >> 6;
      const long long w0 = (long long)offs[pid] >> 6;
      piece[(long long)pid * cap + (w - w0)] = val;
    }
  }
}

extern "C" __global__ void __launch_bounds__(256)
eval_finish_kernel(const float* __restrict__ piece, const int* __restrict__ offs,
                   const float* __restrict__ labels, long long P,
                   long long p_split, int cap, float* __restrict__ partials,
                   float* __restrict__ dO, float inv_b) {
  float c0 = 0.f, c1 = 0.f;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long p = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       p < P; p += stride) {
    const int lo = offs[p], hi = offs[p + 1];
    const long long w0 = (long long)lo >> 6;
    const long long w1 = ((long long)hi - 1) >> 6;
    float o = 0.f;
    const float* pp = piece + (long long)p * cap;
    for (long long w = w0; w <= w1; ++w) o += pp[w - w0];
    const float y = labels[p];
    const float corr = (((o > 0.f ? 1.f : 0.f) == y) ? 1.f : 0.f);
    if (p < p_split) {
      c0 += corr;
      if (dO) dO[p] = (1.f / (1.f + expf(-o)) - y) * inv_b;
    } else {
      c1 += corr;
    }
  }
  c0 = wave_sum(c0);
  c1 = wave_sum(c1);
  __shared__ float sm[8];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wib = threadIdx.x >> 6;
  if (lane == 0) { sm[wib] = c0; sm[4 + wib] = c1; }
  __syncthreads();
  if (threadIdx.x == 0) {
    partials[2 * blockIdx.x] = sm[0] + sm[1] + sm[2] + sm[3];
    partials[2 * blockIdx.x + 1] = sm[4] + sm[5] + sm[6] + sm[7];
  }
}
