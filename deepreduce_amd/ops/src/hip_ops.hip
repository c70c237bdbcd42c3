// deepreduce_amd HIP/CDNA4 kernel library (gfx950-native).
//
// Replaces the reference's composite-op GPU paths (SURVEY.md sect. 2.3 GPU
// kernel inventory): the 1 GB hash-table gather Bloom
// (/root/reference/pytorch/deepreduce.py:431-492) becomes in-register
// MurmurHash3 double hashing; cupy packbits becomes the packed-word wire
// format written directly by the insert kernel; the python-loop QSGD
// (:852-907) becomes one fused block-per-bucket kernel.
//
// Determinism contract: the hash math here is bit-identical to
// deepreduce_amd/hashing.py (fmix32 + Kirsch-Mitzenmacher, 64-bit j*h2
// accumulate before mod) — parity-tested in tests/test_gpu_parity.py.
//
// Wire bit order: bit b -> byte b>>3, bit b&7 (LSB-first); equivalently
// little-endian uint32 word b>>5, bit b&31.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <algorithm>
#include <vector>

#define WAVE 64
#define CHECK_CUDA(x) TORCH_CHECK(x.is_cuda(), #x " must be a GPU tensor")

static inline int64_t ceil_div(int64_t a, int64_t b) { return (a + b - 1) / b; }

// independent-load head width for the universe query (see bt_qcount):
// runtime-tunable via DEEPREDUCE_QHEAD (1..4).  MEASURED same-box
// (gpurun_out/qcount_head_sweep.json, 40-iter A/B): whole-model compress
// 0.499/0.512/0.562 ms and R=8 interleaved decode 1.629/1.589/1.950 ms
// at head 1/2/4 — two independent probe loads balance chain-breaking
// against wasted loads (fill=0.5 kills half the candidates at probe 1),
// so the default is 2.
static inline int qhead_env() {
    static int v = -1;
    if (v < 0) {
        const char* e = getenv("DEEPREDUCE_QHEAD");
        v = e ? atoi(e) : 2;
        if (v < 1) v = 1;
        if (v > 4) v = 4;
    }
    return v;
}

// ---------------------------------------------------------------------------
// hashing (must mirror deepreduce_amd/hashing.py exactly)
// ---------------------------------------------------------------------------

__host__ __device__ __forceinline__ uint32_t fmix32(uint32_t h) {
    h ^= h >> 16;
    h *= 0x85EBCA6Bu;
    h ^= h >> 13;
    h *= 0xC2B2AE35u;
    h ^= h >> 16;
    return h;
}

#define H2_SALT 0x6B43A9B5u

__host__ __device__ __forceinline__ void hash_bases(int64_t item, uint32_t* h1, uint32_t* h2) {
    uint32_t x = (uint32_t)(item & 0xFFFFFFFFll);
    *h1 = fmix32(x + 1u);
    *h2 = fmix32(*h1 ^ H2_SALT) | 1u;
}

// Position derivation: Lemire multiply-shift range reduction
//   pos_j = ((h1 + j*h2 mod 2^32) * m) >> 32
// instead of (h1 + j*h2) mod m.  Uniform on [0, m) for m < 2^31 and
// replaces the two 64-bit integer divisions per candidate (the dominant
// VALU cost of the full-universe query) with one 32x32 multiply-high per
// probe.  Bit-identical to hashing.py's bloom_positions (parity-tested).
__host__ __device__ __forceinline__ uint64_t bloom_pos(uint32_t x, int64_t m) {
    return ((uint64_t)x * (uint64_t)m) >> 32;
}

__host__ __device__ __forceinline__ bool bloom_test(const uint8_t* __restrict__ bits, int64_t m,
                                           int k, int64_t item) {
    uint32_t h1, h2;
    hash_bases(item, &h1, &h2);
    uint32_t x = h1;
    for (int j = 0;;) {
        uint64_t pos = bloom_pos(x, m);
        if (!((bits[pos >> 3] >> (pos & 7)) & 1)) return false;
        if (++j >= k) return true;
        x += h2;
    }
}

// ---------------------------------------------------------------------------
// bloom insert: race-tolerant atomicOr bit sets (idempotent — paper App. E)
// ---------------------------------------------------------------------------

// Graph-safe zero fill (hipMemsetAsync becomes a memset NODE under stream
// capture; a plain kernel keeps the captured graph homogeneous)
__global__ void fill_zero_kernel(int* __restrict__ p, int64_t n) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) p[i] = 0;
}
static inline void zero_ints(int* p, int64_t n, hipStream_t s) {
    int blocks = (int)min((long long)ceil_div(n, 256), 1024ll);
    hipLaunchKernelGGL(fill_zero_kernel, dim3(blocks), dim3(256), 0, s, p, n);
}

__global__ void bloom_insert_kernel(const int64_t* __restrict__ items, int64_t n, int k,
                                    int64_t m, uint32_t* __restrict__ words) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        uint32_t h1, h2;
        hash_bases(items[i], &h1, &h2);
        uint32_t x = h1;
        for (int j = 0; j < k; ++j, x += h2) {
            int64_t pos = (int64_t)bloom_pos(x, m);
            atomicOr(&words[pos >> 5], 1u << (pos & 31));
        }
    }
}

torch::Tensor bloom_insert(torch::Tensor idxs, int64_t m, int64_t num_hash) {
    CHECK_CUDA(idxs);
    auto items = idxs.to(torch::kInt64).contiguous();
    int64_t nwords = ceil_div(m, 32);
    auto words = torch::empty({nwords}, torch::dtype(torch::kInt32).device(idxs.device()));
    int64_t n = items.numel();
    hipStream_t stream = at::hip::getCurrentHIPStream();
    zero_ints(words.data_ptr<int32_t>(), nwords, stream);
    if (n > 0) {
        int threads = 256;
        int blocks = (int)std::min<int64_t>(ceil_div(n, threads), 4096);
        hipLaunchKernelGGL(bloom_insert_kernel, dim3(blocks), dim3(threads), 0, stream,
                           items.data_ptr<int64_t>(), n, (int)num_hash, m,
                           (uint32_t*)words.data_ptr<int32_t>());
    }
    // narrow at offset 0 of a contiguous 1-D tensor is itself contiguous: a
    // VIEW, not a copy
    return words.view(torch::kUInt8).narrow(0, 0, ceil_div(m, 8));
}

// ---------------------------------------------------------------------------
// bloom query + ordered stream compaction (the hot decompress kernel):
// two-pass deterministic — per-block count, torch cumsum, ordered scatter
// via wave ballot + LDS cross-wave prefix.  O(d*k/64) wave-ops total.
// ---------------------------------------------------------------------------

#define QBLOCK 256
#define MAXR 16  // max ranks per batched query

// chunk sizing: enough blocks to fill 256 CUs several times over, chunk a
// multiple of the block size
static inline int64_t query_chunk(int64_t universe) {
    int64_t chunk = ceil_div(universe, 2048);
    chunk = ceil_div(chunk, QBLOCK) * QBLOCK;
    return std::max<int64_t>(chunk, QBLOCK);
}

// Pass 1: test the filter(s), emit (a) per-block per-rank counts and (b) a
// predicate bit-plane per rank (one uint64 ballot word per wave) so pass 2
// never re-hashes.  R filters share the SAME probe positions (the hash does
// not depend on the filter), so hashing is amortized R-fold.
__global__ void bloom_count_kernel(const uint8_t* __restrict__ bits, int64_t stride_bytes,
                                   int R, int64_t m, int k, int64_t universe, int64_t chunk,
                                   int* __restrict__ block_counts /*[R, nblocks]*/,
                                   uint64_t* __restrict__ mask /*[R, ceil(u/64)]*/,
                                   int64_t mask_stride) {
    int64_t start = (int64_t)blockIdx.x * chunk;
    int64_t end = min(start + chunk, universe);
    int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    int cnt[MAXR];
    for (int r = 0; r < R; ++r) cnt[r] = 0;

    for (int64_t i0 = start; i0 < end; i0 += blockDim.x) {
        int64_t i = i0 + threadIdx.x;
        unsigned alive = (i < end) ? ((1u << R) - 1) : 0u;
        if (alive) {
            uint32_t h1, h2;
            hash_bases(i, &h1, &h2);
            uint32_t x = h1;
            for (int j = 0; j < k && alive; ++j, x += h2) {
                uint64_t pos = bloom_pos(x, m);
                int64_t byte = pos >> 3;
                uint8_t bit = pos & 7;
                for (int r = 0; r < R; ++r)
                    if (alive & (1u << r))
                        if (!((bits[r * stride_bytes + byte] >> bit) & 1)) alive &= ~(1u << r);
            }
        }
        for (int r = 0; r < R; ++r) {
            bool pred = alive & (1u << r);
            uint64_t ball = __ballot(pred);
            cnt[r] += __popcll(ball);
            if (lane == 0) mask[r * mask_stride + ((i0 + (int64_t)wid * WAVE) >> 6)] = ball;
        }
    }
    // reduce counts across the block (cnt[r] is wave-uniform after popcll)
    __shared__ int wsum[MAXR][QBLOCK / WAVE];
    if (lane == 0)
        for (int r = 0; r < R; ++r) wsum[r][wid] = cnt[r];
    __syncthreads();
    if (threadIdx.x == 0) {
        for (int r = 0; r < R; ++r) {
            int total = 0;
            for (int w = 0; w < QBLOCK / WAVE; ++w) total += wsum[r][w];
            block_counts[r * gridDim.x + blockIdx.x] = total;
        }
    }
}

// Exclusive scan along dim 1 of an int32 [R, n] matrix, one block per row.
// Replaces the cumsum/sub/to/contiguous torch-op chain in the query and
// top-k drivers (the launch-bound hot path: SURVEY.md sect. 3.1 note).
__global__ void exclusive_scan_rows_kernel(const int* __restrict__ in, int64_t n,
                                           int* __restrict__ out,
                                           int* __restrict__ row_totals /*nullable [R]*/) {
    const int64_t row = blockIdx.x;
    const int* src = in + row * n;
    int* dst = out + row * n;
    int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    __shared__ int wave_tot[QBLOCK / WAVE];
    __shared__ int carry;
    if (threadIdx.x == 0) carry = 0;
    __syncthreads();
    for (int64_t i0 = 0; i0 < n; i0 += blockDim.x) {
        int64_t i = i0 + threadIdx.x;
        int v = (i < n) ? src[i] : 0;
        // inclusive wave scan
        int incl = v;
        for (int off = 1; off < WAVE; off <<= 1) {
            int up = __shfl_up(incl, off, WAVE);
            if (lane >= off) incl += up;
        }
        if (lane == WAVE - 1) wave_tot[wid] = incl;
        __syncthreads();
        int wbase = 0;
        for (int w = 0; w < wid; ++w) wbase += wave_tot[w];
        if (i < n) dst[i] = carry + wbase + incl - v;
        __syncthreads();
        if (threadIdx.x == 0) {
            int t = 0;
            for (int w = 0; w < QBLOCK / WAVE; ++w) t += wave_tot[w];
            carry += t;
        }
        __syncthreads();
    }
    if (row_totals != nullptr && threadIdx.x == 0) row_totals[row] = carry;
}

// Pass 2: ordered compaction of the predicate bit-plane (no hashing).
// rank_base: per-rank output base — explicit device array, or r*rank_stride
// when rank_base == nullptr (keeps the sync-free leftmost path free of
// torch-op glue).
__global__ void bloom_scatter_kernel(const uint64_t* __restrict__ mask, int64_t mask_stride,
                                     int R, int64_t universe, int64_t chunk,
                                     const int* __restrict__ block_offsets /*[R, nblocks]*/,
                                     const int64_t* __restrict__ rank_base /*[R] or null*/,
                                     int64_t rank_stride,
                                     int64_t max_per_rank /* <=0: unlimited */,
                                     int64_t* __restrict__ out) {
    int64_t start = (int64_t)blockIdx.x * chunk;
    int64_t end = min(start + chunk, universe);
    int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    __shared__ int wave_cnt[MAXR][QBLOCK / WAVE];
    __shared__ int base_s[MAXR];
    __shared__ int64_t rbase_s[MAXR];
    if (threadIdx.x < MAXR && threadIdx.x < R) {
        rbase_s[threadIdx.x] =
            rank_base ? rank_base[threadIdx.x] : (int64_t)threadIdx.x * rank_stride;
        base_s[threadIdx.x] =
            block_offsets[threadIdx.x * gridDim.x + blockIdx.x] + (int)rbase_s[threadIdx.x];
    }
    __syncthreads();
    for (int64_t i0 = start; i0 < end; i0 += blockDim.x) {
        int64_t i = i0 + threadIdx.x;
        int64_t w64 = (i0 + (int64_t)wid * WAVE) >> 6;
        for (int r = 0; r < R; ++r) {
            uint64_t ball = mask[r * mask_stride + w64];
            if (lane == 0) wave_cnt[r][wid] = __popcll(ball);
        }
        __syncthreads();
        for (int r = 0; r < R; ++r) {
            uint64_t ball = mask[r * mask_stride + w64];
            bool pred = (ball >> lane) & 1;
            if (pred) {
                int wbase = 0;
                for (int w = 0; w < wid; ++w) wbase += wave_cnt[r][w];
                int prefix = __popcll(ball & ((lane == 63) ? ~0ull >> 1 : ((1ull << lane) - 1)));
                int64_t oi = base_s[r] + wbase + prefix;
                if (max_per_rank <= 0 || (oi - rbase_s[r]) < max_per_rank) out[oi] = i;
            }
        }
        __syncthreads();
        if (threadIdx.x < MAXR && threadIdx.x < R) {
            int total = 0;
            for (int w = 0; w < QBLOCK / WAVE; ++w) total += wave_cnt[threadIdx.x][w];
            base_s[threadIdx.x] += total;
        }
        __syncthreads();
    }
}

// shared driver: R stacked filters -> (positives flat [sum], counts [R])
static std::vector<torch::Tensor> query_multi_impl(torch::Tensor bits2d, int64_t m,
                                                   int64_t num_hash, int64_t universe) {
    auto bits = bits2d.contiguous();
    int R = (int)bits.size(0);
    TORCH_CHECK(R >= 1 && R <= MAXR, "1..16 ranks supported");
    int64_t stride_bytes = bits.size(1);
    int64_t chunk = query_chunk(universe);
    int64_t nblocks = ceil_div(universe, chunk);
    auto dev = bits.device();
    auto counts = torch::empty({R, nblocks}, torch::dtype(torch::kInt32).device(dev));
    // mask rows padded so every wave's ballot word has a slot
    int64_t mask_words = ceil_div(nblocks * chunk, 64);
    auto mask = torch::empty({R, mask_words}, torch::dtype(torch::kInt64).device(dev));
    hipStream_t stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(bloom_count_kernel, dim3((int)nblocks), dim3(QBLOCK), 0, stream,
                       bits.data_ptr<uint8_t>(), stride_bytes, R, m, (int)num_hash, universe,
                       chunk, counts.data_ptr<int>(), (uint64_t*)mask.data_ptr<int64_t>(),
                       mask_words);
    auto offsets = torch::empty({R, nblocks}, torch::dtype(torch::kInt32).device(dev));
    auto row_totals = torch::empty({R}, torch::dtype(torch::kInt32).device(dev));
    hipLaunchKernelGGL(exclusive_scan_rows_kernel, dim3(R), dim3(QBLOCK), 0, stream,
                       counts.data_ptr<int>(), nblocks, offsets.data_ptr<int>(),
                       row_totals.data_ptr<int>());
    auto rank_totals = row_totals.to(torch::kInt64);
    auto rank_base = rank_totals.cumsum(0) - rank_totals;
    int64_t total = (int64_t)rank_totals.sum().item<int64_t>();  // one sync (ragged path)
    auto out = torch::empty({total}, torch::dtype(torch::kInt64).device(dev));
    if (total > 0) {
        auto rank_base_c = rank_base.contiguous();
        hipLaunchKernelGGL(bloom_scatter_kernel, dim3((int)nblocks), dim3(QBLOCK), 0, stream,
                           (const uint64_t*)mask.data_ptr<int64_t>(), mask_words, R, universe,
                           chunk, offsets.data_ptr<int>(), rank_base_c.data_ptr<int64_t>(),
                           (int64_t)0, (int64_t)0, out.data_ptr<int64_t>());
    }
    return {out, rank_totals};
}

// Sync-free leftmost-k query: returns the FIRST k positives per rank as
// int64 [R, k] with NO host round-trip (output size is host-known).
// Invariant: a Bloom filter has no false negatives, so with k <= number of
// inserted distinct items every row is fully populated; rows are ascending.
torch::Tensor bloom_query_leftmost(torch::Tensor packed2d, int64_t m, int64_t num_hash,
                                   int64_t universe, int64_t k_out) {
    CHECK_CUDA(packed2d);
    auto bits = packed2d.dim() == 1 ? packed2d.unsqueeze(0).contiguous() : packed2d.contiguous();
    int R = (int)bits.size(0);
    TORCH_CHECK(R >= 1 && R <= MAXR, "1..16 ranks supported");
    int64_t stride_bytes = bits.size(1);
    int64_t chunk = query_chunk(universe);
    int64_t nblocks = ceil_div(universe, chunk);
    auto dev = bits.device();
    // one int32 workspace: counts [R, nblocks] | offsets [R, nblocks]
    auto ws = torch::empty({2 * R * nblocks}, torch::dtype(torch::kInt32).device(dev));
    int* counts = ws.data_ptr<int>();
    int* offsets = counts + R * nblocks;
    int64_t mask_words = ceil_div(nblocks * chunk, 64);
    auto mask = torch::empty({R, mask_words}, torch::dtype(torch::kInt64).device(dev));
    hipStream_t stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(bloom_count_kernel, dim3((int)nblocks), dim3(QBLOCK), 0, stream,
                       bits.data_ptr<uint8_t>(), stride_bytes, R, m, (int)num_hash, universe,
                       chunk, counts, (uint64_t*)mask.data_ptr<int64_t>(),
                       mask_words);
    hipLaunchKernelGGL(exclusive_scan_rows_kernel, dim3(R), dim3(QBLOCK), 0, stream,
                       counts, nblocks, offsets, (int*)nullptr);
    auto out = torch::empty({R * k_out}, torch::dtype(torch::kInt64).device(dev));
    zero_ints((int*)out.data_ptr<int64_t>(), 2 * R * k_out, stream);
    hipLaunchKernelGGL(bloom_scatter_kernel, dim3((int)nblocks), dim3(QBLOCK), 0, stream,
                       (const uint64_t*)mask.data_ptr<int64_t>(), mask_words, R, universe,
                       chunk, offsets, (const int64_t*)nullptr, k_out, k_out,
                       out.data_ptr<int64_t>());
    return out.view({R, k_out});
}

torch::Tensor bloom_query_positives(torch::Tensor packed, int64_t m, int64_t num_hash,
                                    int64_t universe) {
    CHECK_CUDA(packed);
    auto res = query_multi_impl(packed.contiguous().unsqueeze(0), m, num_hash, universe);
    return res[0];
}

// batched: packed2d [R, nbytes] (one row per rank, same m/k/universe) ->
// (positives concatenated rank-major, per-rank counts)
std::vector<torch::Tensor> bloom_query_positives_multi(torch::Tensor packed2d, int64_t m,
                                                       int64_t num_hash, int64_t universe) {
    CHECK_CUDA(packed2d);
    TORCH_CHECK(packed2d.dim() == 2, "expected [R, nbytes]");
    return query_multi_impl(packed2d, m, num_hash, universe);
}

__global__ void bloom_members_kernel(const uint8_t* __restrict__ bits, int64_t m, int k,
                                     const int64_t* __restrict__ items, int64_t n,
                                     bool* __restrict__ out) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) out[i] = bloom_test(bits, m, k, items[i]);
}

torch::Tensor bloom_query_members(torch::Tensor packed, int64_t m, int64_t num_hash,
                                  torch::Tensor items) {
    CHECK_CUDA(packed);
    auto bits = packed.contiguous();
    auto it = items.to(torch::kInt64).contiguous();
    int64_t n = it.numel();
    auto out = torch::empty({n}, torch::dtype(torch::kBool).device(bits.device()));
    if (n > 0) {
        int threads = 256;
        int blocks = (int)std::min<int64_t>(ceil_div(n, threads), 4096);
        hipStream_t stream = at::hip::getCurrentHIPStream();
        hipLaunchKernelGGL(bloom_members_kernel, dim3(blocks), dim3(threads), 0, stream,
                           bits.data_ptr<uint8_t>(), m, (int)num_hash, it.data_ptr<int64_t>(),
                           n, out.data_ptr<bool>());
    }
    return out;
}

// ---------------------------------------------------------------------------
// QSGD: fused per-bucket norm + stochastic quantize (one block per bucket)
// ---------------------------------------------------------------------------

__global__ void qsgd_quantize_kernel(const float* __restrict__ v, int64_t n, int bucket,
                                     float quantum, uint64_t seed,
                                     int8_t* __restrict__ out_levels,
                                     float* __restrict__ out_norms) {
    int64_t b = blockIdx.x;
    int64_t start = b * bucket;
    int64_t end = min(start + (int64_t)bucket, n);
    float ss = 0.f;
    for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x) {
        float x = v[i];
        ss += x * x;
    }
    for (int off = WAVE / 2; off > 0; off >>= 1) ss += __shfl_down(ss, off, WAVE);
    __shared__ float wsum[256 / WAVE];
    __shared__ float norm_s;
    int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    if (lane == 0) wsum[wid] = ss;
    __syncthreads();
    if (threadIdx.x == 0) {
        float t = 0.f;
        for (int w = 0; w < (int)(blockDim.x / WAVE); ++w) t += wsum[w];
        norm_s = sqrtf(t);
        out_norms[b] = norm_s;
    }
    __syncthreads();
    float norm = norm_s;
    float scale = (norm > 0.f) ? quantum / norm : 0.f;
    for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x) {
        float x = v[i];
        float lf = scale * fabsf(x);
        float prev = floorf(lf);
        // stateless uniform in [0,1): fmix of (seed, i)
        uint32_t r = fmix32((uint32_t)(i & 0xFFFFFFFF) ^ fmix32((uint32_t)(seed & 0xFFFFFFFF)));
        float u = (float)r * (1.0f / 4294967296.0f);
        float lvl = prev + ((u < (lf - prev)) ? 1.f : 0.f);
        float sgn = (x > 0.f) ? 1.f : ((x < 0.f) ? -1.f : 0.f);
        out_levels[i] = (int8_t)(lvl * sgn);
    }
}

std::vector<torch::Tensor> qsgd_quantize(torch::Tensor vals, int64_t quantum_num,
                                         int64_t bucket_size) {
    CHECK_CUDA(vals);
    auto v = vals.to(torch::kFloat32).contiguous();
    int64_t n = v.numel();
    int64_t nb = ceil_div(n, bucket_size);
    auto levels = torch::empty({n}, torch::dtype(torch::kInt8).device(v.device()));
    auto norms = torch::empty({nb}, torch::dtype(torch::kFloat32).device(v.device()));
    if (n > 0) {
        uint64_t seed = (uint64_t)torch::randint(0, 1 << 30, {1}).item<int64_t>();
        hipStream_t stream = at::hip::getCurrentHIPStream();
        hipLaunchKernelGGL(qsgd_quantize_kernel, dim3((int)nb), dim3(256), 0, stream,
                           v.data_ptr<float>(), n, (int)bucket_size, (float)quantum_num, seed,
                           levels.data_ptr<int8_t>(), norms.data_ptr<float>());
    }
    return {levels, norms};
}

__global__ void qsgd_dequantize_kernel(const int8_t* __restrict__ levels,
                                       const float* __restrict__ norms, int64_t n, int bucket,
                                       float inv_quantum, float* __restrict__ out) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) out[i] = norms[i / bucket] * inv_quantum * (float)levels[i];
}

torch::Tensor qsgd_dequantize(torch::Tensor levels, torch::Tensor norms, int64_t quantum_num,
                              int64_t bucket_size) {
    CHECK_CUDA(levels);
    auto l = levels.contiguous();
    auto nm = norms.to(torch::kFloat32).contiguous();
    int64_t n = l.numel();
    auto out = torch::empty({n}, torch::dtype(torch::kFloat32).device(l.device()));
    if (n > 0) {
        int threads = 256;
        int blocks = (int)std::min<int64_t>(ceil_div(n, threads), 4096);
        hipStream_t stream = at::hip::getCurrentHIPStream();
        hipLaunchKernelGGL(qsgd_dequantize_kernel, dim3(blocks), dim3(threads), 0, stream,
                           l.data_ptr<int8_t>(), nm.data_ptr<float>(), n, (int)bucket_size,
                           1.0f / (float)quantum_num, out.data_ptr<float>());
    }
    return out;
}

// ---------------------------------------------------------------------------
// n-bit integer pack/unpack (LSB-first bitstream; mirrors ops/reference.py)
// ---------------------------------------------------------------------------

__global__ void pack_ints_kernel(const int64_t* __restrict__ v, int64_t n, int nbits,
                                 uint32_t* __restrict__ words, int64_t nwords) {
    int64_t w = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; w < nwords; w += stride) {
        uint32_t acc = 0;
        int64_t bit0 = w * 32;
        for (int b = 0; b < 32; ++b) {
            int64_t bit = bit0 + b;
            int64_t i = bit / nbits;
            if (i >= n) break;
            int j = (int)(bit - i * nbits);
            acc |= (uint32_t)((v[i] >> j) & 1) << b;
        }
        words[w] = acc;
    }
}

torch::Tensor pack_ints(torch::Tensor values, int64_t nbits) {
    CHECK_CUDA(values);
    auto v = values.to(torch::kInt64).contiguous();
    int64_t n = v.numel();
    int64_t total_bits = n * nbits;
    int64_t nbytes = ceil_div(total_bits, 8);
    int64_t nwords = ceil_div(total_bits, 32);
    auto words = torch::zeros({std::max<int64_t>(nwords, 1)},
                              torch::dtype(torch::kInt32).device(v.device()));
    if (n > 0) {
        int threads = 256;
        int blocks = (int)std::min<int64_t>(ceil_div(nwords, threads), 4096);
        hipStream_t stream = at::hip::getCurrentHIPStream();
        hipLaunchKernelGGL(pack_ints_kernel, dim3(blocks), dim3(threads), 0, stream,
                           v.data_ptr<int64_t>(), n, (int)nbits,
                           (uint32_t*)words.data_ptr<int32_t>(), nwords);
    }
    return words.view(torch::kUInt8).narrow(0, 0, nbytes).contiguous();
}

__global__ void unpack_ints_kernel(const uint8_t* __restrict__ stream, int64_t n, int nbits,
                                   int64_t* __restrict__ out) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        int64_t bit0 = i * nbits;
        int64_t byte0 = bit0 >> 3;
        int off = (int)(bit0 & 7);
        uint64_t acc = 0;
        int need = (nbits + off + 7) / 8;
        for (int b = 0; b < need; ++b) acc |= (uint64_t)stream[byte0 + b] << (8 * b);
        out[i] = (int64_t)((acc >> off) & ((nbits == 64) ? ~0ull : ((1ull << nbits) - 1)));
    }
}

torch::Tensor unpack_ints(torch::Tensor stream, int64_t n, int64_t nbits) {
    CHECK_CUDA(stream);
    auto s = stream.contiguous();
    auto out = torch::empty({n}, torch::dtype(torch::kInt64).device(s.device()));
    if (n > 0) {
        int threads = 256;
        int blocks = (int)std::min<int64_t>(ceil_div(n, threads), 4096);
        hipStream_t stream_ = at::hip::getCurrentHIPStream();
        hipLaunchKernelGGL(unpack_ints_kernel, dim3(blocks), dim3(threads), 0, stream_,
                           s.data_ptr<uint8_t>(), n, (int)nbits, out.data_ptr<int64_t>());
    }
    return out;
}


// ---------------------------------------------------------------------------
// Deterministic histogram-threshold top-k select (replaces torch.topk's
// sort-based path: 3 kernels, no sort, no host sync).
//
// |x| as IEEE-754 bits is monotonic for non-negative floats, so top-k by
// magnitude == top-k by the uint32 key bits(|x|).  An 11-bit histogram of
// key>>20 (2048 bins) locates the threshold bin; elements in strictly
// higher bins are all selected, and the tie-bin contributes its LEFTMOST
// (lowest-index) remainder — fully deterministic.
// ---------------------------------------------------------------------------

#define TK_BINS 2048
#define TK_BLOCK 256

// 22-bit monotonic magnitude key: |x| as IEEE bits >> 9.  Two 11-bit radix
// levels; remaining ties (low 9 mantissa bits, < 0.012% relative) resolve
// deterministically to the LEFTMOST index.
__device__ __forceinline__ uint32_t tk_key22(float x) {
    union { float f; uint32_t u; } c;
    c.f = fabsf(x);
    return c.u >> 9;
}

__global__ void topk_hist_kernel(const float* __restrict__ v, int64_t n,
                                 int* __restrict__ hist) {
    __shared__ int lh[TK_BINS];
    for (int b = threadIdx.x; b < TK_BINS; b += blockDim.x) lh[b] = 0;
    __syncthreads();
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) atomicAdd(&lh[tk_key22(v[i]) >> 11], 1);
    __syncthreads();
    for (int b = threadIdx.x; b < TK_BINS; b += blockDim.x)
        if (lh[b]) atomicAdd(&hist[b], lh[b]);
}

// level 2: histogram of the LOW 11 key bits, only for elements whose high
// bits equal bstar
__global__ void topk_hist2_kernel(const float* __restrict__ v, int64_t n,
                                  const int* __restrict__ bstar_p,
                                  int* __restrict__ hist) {
    const uint32_t bstar = (uint32_t)*bstar_p;
    __shared__ int lh[TK_BINS];
    for (int b = threadIdx.x; b < TK_BINS; b += blockDim.x) lh[b] = 0;
    __syncthreads();
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        uint32_t key = tk_key22(v[i]);
        if ((key >> 11) == bstar) atomicAdd(&lh[key & 2047u], 1);
    }
    __syncthreads();
    for (int b = threadIdx.x; b < TK_BINS; b += blockDim.x)
        if (lh[b]) atomicAdd(&hist[b], lh[b]);
}

// Derive the threshold bin from a 2048-bin histogram on-device (one
// wavefront): bstar = largest bin b with suffix_sum(b) >= target, above =
// suffix_sum(bstar+1).  Replaces the flip/cumsum/cat/index torch-op chain
// that made the old driver launch-bound (~20 composite ops per tensor).
// scalars layout: [0]=bstar1 [1]=above1 [2]=thresh22 [3]=count_above.
// level 1: target = k.  level 2: target = k - above1; also finalizes
// thresh22 = bstar1*2048 + bstar2 and count_above = above1 + above2.
__global__ void topk_thresh_kernel(const int* __restrict__ hist, int64_t k, int level,
                                   int* __restrict__ sc) {
    const int lane = threadIdx.x;               // single wave of 64
    const int target = (level == 1) ? (int)k : (int)k - sc[1];
    const int SEG = TK_BINS / WAVE;             // 32 bins per lane
    int h[SEG];
    int seg_sum = 0;
    for (int j = 0; j < SEG; ++j) {
        h[j] = hist[lane * SEG + j];
        seg_sum += h[j];
    }
    // suffix over lanes: sum of seg_sum for lanes strictly greater
    int suffix_excl = 0;
    {
        int acc = seg_sum;
        for (int off = 1; off < WAVE; off <<= 1) {
            int other = __shfl_down(acc, off, WAVE);
            if (lane + off < WAVE) acc += other;
        }
        // acc now = suffix inclusive at this lane; recover exclusive
        suffix_excl = acc - seg_sum;
    }
    // walk own segment from the right: first j (largest) with suffix >= target
    int found_b = -1, found_above = 0;
    int acc = 0;
    for (int j = SEG - 1; j >= 0; --j) {
        int suffix_here = suffix_excl + acc + h[j];
        if (suffix_here >= target) {
            found_b = lane * SEG + j;
            found_above = suffix_excl + acc;  // suffix(b+1)
            break;
        }
        acc += h[j];
    }
    // max-reduce found_b across the wave; winning lane writes results
    int best = found_b;
    for (int off = 1; off < WAVE; off <<= 1) {
        int other = __shfl_down(best, off, WAVE);
        if (lane + off < WAVE && other > best) best = other;
    }
    best = __shfl(best, 0, WAVE);
    if (found_b == best && found_b >= 0) {
        if (level == 1) {
            sc[0] = found_b;
            sc[1] = found_above;
        } else {
            sc[2] = sc[0] * TK_BINS + found_b;
            sc[3] = sc[1] + found_above;
        }
    }
}

// per-block counts of {key22 > T} and {key22 == T}
__global__ void topk_count_kernel(const float* __restrict__ v, int64_t n, int64_t chunk,
                                  const int* __restrict__ thresh_p,
                                  int* __restrict__ counts /*[2, nblocks]*/) {
    const uint32_t T = (uint32_t)*thresh_p;
    int64_t start = (int64_t)blockIdx.x * chunk;
    int64_t end = min(start + chunk, n);
    int c0 = 0, c1 = 0;
    for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x) {
        uint32_t key = tk_key22(v[i]);
        c0 += (key > T);
        c1 += (key == T);
    }
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        c0 += __shfl_down(c0, off, WAVE);
        c1 += __shfl_down(c1, off, WAVE);
    }
    __shared__ int w0[TK_BLOCK / WAVE], w1[TK_BLOCK / WAVE];
    int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    if (lane == 0) { w0[wid] = c0; w1[wid] = c1; }
    __syncthreads();
    if (threadIdx.x == 0) {
        int t0 = 0, t1 = 0;
        for (int w = 0; w < TK_BLOCK / WAVE; ++w) { t0 += w0[w]; t1 += w1[w]; }
        counts[blockIdx.x] = t0;
        counts[gridDim.x + blockIdx.x] = t1;
    }
}

__global__ void topk_scatter_kernel(const float* __restrict__ v, int64_t n, int64_t chunk,
                                    const int* __restrict__ thresh_p,
                                    const int* __restrict__ off0 /*[nblocks] excl-cum of c0*/,
                                    const int* __restrict__ off1 /*[nblocks] excl-cum of c1*/,
                                    const int* __restrict__ count_above_p,
                                    int64_t k,
                                    float* __restrict__ out_v, int64_t* __restrict__ out_i) {
    const uint32_t T = (uint32_t)*thresh_p;
    const int64_t above = *count_above_p;   // elements with key22 > T
    const int64_t need = k - above;         // taken from the tie class
    int64_t start = (int64_t)blockIdx.x * chunk;
    int64_t end = min(start + chunk, n);
    int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    __shared__ int wc0[TK_BLOCK / WAVE], wc1[TK_BLOCK / WAVE];
    __shared__ int base0_s, base1_s;
    if (threadIdx.x == 0) { base0_s = off0[blockIdx.x]; base1_s = off1[blockIdx.x]; }
    __syncthreads();
    for (int64_t i0 = start; i0 < end; i0 += blockDim.x) {
        int64_t i = i0 + threadIdx.x;
        bool in_r = (i < end);
        uint32_t key = in_r ? tk_key22(v[i]) : 0u;
        bool p0 = in_r && (key > T);
        bool p1 = in_r && (key == T);
        uint64_t b0 = __ballot(p0), b1 = __ballot(p1);
        if (lane == 0) { wc0[wid] = __popcll(b0); wc1[wid] = __popcll(b1); }
        __syncthreads();
        uint64_t below = (lane == 63) ? (~0ull >> 1) : ((1ull << lane) - 1);
        if (p0) {
            int wbase = 0;
            for (int w = 0; w < wid; ++w) wbase += wc0[w];
            int64_t oi = base0_s + wbase + __popcll(b0 & below);
            out_i[oi] = i;
            out_v[oi] = v[i];
        } else if (p1) {
            int wbase = 0;
            for (int w = 0; w < wid; ++w) wbase += wc1[w];
            int64_t ordinal = base1_s + wbase + __popcll(b1 & below);
            if (ordinal < need) {
                out_i[above + ordinal] = i;
                out_v[above + ordinal] = v[i];
            }
        }
        __syncthreads();
        if (threadIdx.x == 0) {
            int t0 = 0, t1 = 0;
            for (int w = 0; w < TK_BLOCK / WAVE; ++w) { t0 += wc0[w]; t1 += wc1[w]; }
            base0_s += t0;
            base1_s += t1;
        }
        __syncthreads();
    }
}

std::vector<torch::Tensor> topk_select(torch::Tensor flat, int64_t k) {
    CHECK_CUDA(flat);
    auto v = flat.to(torch::kFloat32).contiguous();
    int64_t n = v.numel();
    TORCH_CHECK(k >= 1 && k <= n, "k out of range");
    auto dev = v.device();
    hipStream_t stream = at::hip::getCurrentHIPStream();

    // Single int32 workspace; 7 kernels, zero composite torch ops, zero
    // host syncs.  Layout: hist1[2048] | hist2[2048] | scalars[4] |
    // counts[2*nblocks] | offsets[2*nblocks].
    int64_t chunk = query_chunk(n);
    int64_t nblocks = ceil_div(n, chunk);
    auto ws = torch::empty({2 * TK_BINS + 4 + 4 * nblocks},
                           torch::dtype(torch::kInt32).device(dev));
    int* hist1 = ws.data_ptr<int>();
    int* hist2 = hist1 + TK_BINS;
    int* sc = hist2 + TK_BINS;
    int* counts = sc + 4;
    int* offs = counts + 2 * nblocks;
    zero_ints(hist1, 2 * TK_BINS + 4, stream);

    int hblocks = (int)std::min<int64_t>(ceil_div(n, TK_BLOCK * 16), 2048);
    hipLaunchKernelGGL(topk_hist_kernel, dim3(hblocks), dim3(TK_BLOCK), 0, stream,
                       v.data_ptr<float>(), n, hist1);
    hipLaunchKernelGGL(topk_thresh_kernel, dim3(1), dim3(WAVE), 0, stream, hist1, k, 1, sc);
    hipLaunchKernelGGL(topk_hist2_kernel, dim3(hblocks), dim3(TK_BLOCK), 0, stream,
                       v.data_ptr<float>(), n, &sc[0], hist2);
    hipLaunchKernelGGL(topk_thresh_kernel, dim3(1), dim3(WAVE), 0, stream, hist2, k, 2, sc);

    hipLaunchKernelGGL(topk_count_kernel, dim3((int)nblocks), dim3(TK_BLOCK), 0, stream,
                       v.data_ptr<float>(), n, chunk, &sc[2], counts);
    hipLaunchKernelGGL(exclusive_scan_rows_kernel, dim3(2), dim3(QBLOCK), 0, stream,
                       counts, nblocks, offs, (int*)nullptr);

    auto out_v = torch::empty({k}, torch::dtype(torch::kFloat32).device(dev));
    auto out_i = torch::empty({k}, torch::dtype(torch::kInt64).device(dev));
    hipLaunchKernelGGL(topk_scatter_kernel, dim3((int)nblocks), dim3(TK_BLOCK), 0, stream,
                       v.data_ptr<float>(), n, chunk, &sc[2],
                       offs, offs + nblocks, &sc[3], k,
                       out_v.data_ptr<float>(), out_i.data_ptr<int64_t>());
    return {out_v, out_i};
}

// ---------------------------------------------------------------------------
// Batched tiny SPD solve (polyfit normal equations): one thread per system,
// in-register Cholesky for d1 <= 8.  Replaces the reference's per-segment
// CPU inverse round-trip (pytorch/deepreduce.py:331-334) and rocSOLVER's
// small-batch LU (milliseconds per call) with a microsecond kernel.
// ---------------------------------------------------------------------------

#define SOLVE_MAXD 8

__global__ void cholesky_solve_kernel(const double* __restrict__ G /*[S,d,d]*/,
                                      const double* __restrict__ b /*[S,d]*/,
                                      int S, int d, double* __restrict__ x /*[S,d]*/) {
    int s = blockIdx.x * blockDim.x + threadIdx.x;
    if (s >= S) return;
    double L[SOLVE_MAXD][SOLVE_MAXD];
    double y[SOLVE_MAXD];
    const double* g = G + (int64_t)s * d * d;
    const double* bb = b + (int64_t)s * d;
    // Cholesky G = L L^T (lower)
    for (int i = 0; i < d; ++i) {
        for (int j = 0; j <= i; ++j) {
            double sum = g[i * d + j];
            for (int p = 0; p < j; ++p) sum -= L[i][p] * L[j][p];
            if (i == j) {
                L[i][j] = sqrt(sum > 1e-300 ? sum : 1e-300);
            } else {
                L[i][j] = sum / L[j][j];
            }
        }
    }
    // forward: L y = b
    for (int i = 0; i < d; ++i) {
        double sum = bb[i];
        for (int p = 0; p < i; ++p) sum -= L[i][p] * y[p];
        y[i] = sum / L[i][i];
    }
    // backward: L^T x = y
    double* xx = x + (int64_t)s * d;
    for (int i = d - 1; i >= 0; --i) {
        double sum = y[i];
        for (int p = i + 1; p < d; ++p) sum -= L[p][i] * xx[p];
        xx[i] = sum / L[i][i];
    }
}

torch::Tensor cholesky_solve_small(torch::Tensor G, torch::Tensor b) {
    CHECK_CUDA(G);
    auto g = G.to(torch::kFloat64).contiguous();
    auto bb = b.to(torch::kFloat64).contiguous();
    int S = (int)g.size(0);
    int d = (int)g.size(1);
    TORCH_CHECK(d <= SOLVE_MAXD, "d <= 8");
    auto x = torch::empty({S, d}, g.options());
    if (S > 0) {
        int threads = 64;
        int blocks = (S + threads - 1) / threads;
        hipStream_t stream = at::hip::getCurrentHIPStream();
        hipLaunchKernelGGL(cholesky_solve_kernel, dim3(blocks), dim3(threads), 0, stream,
                           g.data_ptr<double>(), bb.data_ptr<double>(), S, d,
                           x.data_ptr<double>());
    }
    return x;
}


// ===========================================================================
// Batched cross-tensor pipeline (the MI355X-first hot path).
//
// One training step compresses ~54 large tensors.  Launched per tensor, the
// radix-select/bloom kernels are each a few microseconds of real work — the
// step becomes kernel-count bound even under hipGraph replay (profiles/
// NOTES.md r03).  Here the WHOLE model is processed by ~17 kernels total:
// every kernel walks a block->tensor map (b2t) so all tensors' chunks fill
// the 256 CUs concurrently, and the compress side writes the fused wire
// buffer (the exact _flatten_payload layout) directly — no torch-op glue,
// no cats, no per-tensor launches.
//
// Descriptor row (int64 [T, 12], built once per model layout in
// deepreduce_amd/ops/batched.py and cached on device):
//   0 n        elements in tensor
//   1 voff     offset into the flat values buffer
//   2 k        top-k count
//   3 koff     prefix sum of k (output offset)
//   4 m        bloom bits
//   5 nh       bloom hash count
//   6 bitoff   absolute byte offset of the bloom bit array in the wire
//   7 valoff   absolute byte offset of the FP-aware values in the wire
//   8 cntoff   prefix sum of per-tensor block counts (count/offset slots)
//   9 mwoff    prefix sum of per-tensor ballot-mask words
//  10 blkoff   prefix sum of block indices (global block -> local block)
//  11 reserved
// ===========================================================================

#define BT_CHUNK 8192  // elements per block (multiple of 256)
#define BT_F 16

__device__ __forceinline__ const int64_t* bt_row(const int64_t* desc, int t) {
    return desc + (int64_t)t * BT_F;
}

// level 1: 11-bit high-key histogram per tensor; level 2: low 11 bits of the
// level-1 tie bin.
__global__ void bt_hist_kernel(const float* __restrict__ vals,
                               const int64_t* __restrict__ desc,
                               const int* __restrict__ b2t, int level,
                               int* __restrict__ hist /*[T,2048] plane*/,
                               const int* __restrict__ sc /*[T,4]*/) {
    const int t = b2t[blockIdx.x];
    const int64_t* D = bt_row(desc, t);
    const int64_t lb = blockIdx.x - D[10];
    const int64_t start = lb * BT_CHUNK;
    const int64_t end = min(start + BT_CHUNK, D[0]);
    const float* __restrict__ v = vals + D[1];
    __shared__ int lh[TK_BINS];
    for (int i = threadIdx.x; i < TK_BINS; i += blockDim.x) lh[i] = 0;
    __syncthreads();
    if (level == 1) {
        for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x)
            atomicAdd(&lh[tk_key22(v[i]) >> 11], 1);
    } else {
        const uint32_t bstar = (uint32_t)sc[t * 4 + 0];
        for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x) {
            uint32_t key = tk_key22(v[i]);
            if ((key >> 11) == bstar) atomicAdd(&lh[key & 2047u], 1);
        }
    }
    __syncthreads();
    int* H = hist + (int64_t)t * TK_BINS;
    for (int i = threadIdx.x; i < TK_BINS; i += blockDim.x)
        if (lh[i]) atomicAdd(&H[i], lh[i]);
}

// one wavefront per tensor: derive the threshold bin from the histogram
// (same math as topk_thresh_kernel, batched over T)
__global__ void bt_thresh_kernel(const int* __restrict__ hist,
                                 const int64_t* __restrict__ desc, int level,
                                 int* __restrict__ sc) {
    const int t = blockIdx.x;
    const int lane = threadIdx.x;
    const int64_t k = bt_row(desc, t)[2];
    const int* H = hist + (int64_t)t * TK_BINS;
    int* S = sc + t * 4;
    const int target = (level == 1) ? (int)k : (int)k - S[1];
    const int SEG = TK_BINS / WAVE;
    int h[SEG];
    int seg_sum = 0;
    for (int j = 0; j < SEG; ++j) {
        h[j] = H[lane * SEG + j];
        seg_sum += h[j];
    }
    int suffix_excl;
    {
        int acc = seg_sum;
        for (int off = 1; off < WAVE; off <<= 1) {
            int other = __shfl_down(acc, off, WAVE);
            if (lane + off < WAVE) acc += other;
        }
        suffix_excl = acc - seg_sum;
    }
    int found_b = -1, found_above = 0;
    int acc = 0;
    for (int j = SEG - 1; j >= 0; --j) {
        int suffix_here = suffix_excl + acc + h[j];
        if (suffix_here >= target) {
            found_b = lane * SEG + j;
            found_above = suffix_excl + acc;
            break;
        }
        acc += h[j];
    }
    int best = found_b;
    for (int off = 1; off < WAVE; off <<= 1) {
        int other = __shfl_down(best, off, WAVE);
        if (lane + off < WAVE && other > best) best = other;
    }
    best = __shfl(best, 0, WAVE);
    if (found_b == best && found_b >= 0) {
        if (level == 1) {
            S[0] = found_b;
            S[1] = found_above;
        } else {
            S[2] = S[0] * TK_BINS + found_b;
            S[3] = S[1] + found_above;
        }
    }
}

// per-block {key > T} / {key == T} counts -> counts planes [2, BV]
__global__ void bt_count_kernel(const float* __restrict__ vals,
                                const int64_t* __restrict__ desc,
                                const int* __restrict__ b2t,
                                const int* __restrict__ sc, int64_t BV,
                                int* __restrict__ counts) {
    const int t = b2t[blockIdx.x];
    const int64_t* D = bt_row(desc, t);
    const int64_t lb = blockIdx.x - D[10];
    const int64_t start = lb * BT_CHUNK;
    const int64_t end = min(start + BT_CHUNK, D[0]);
    const float* __restrict__ v = vals + D[1];
    const uint32_t T22 = (uint32_t)sc[t * 4 + 2];
    int c0 = 0, c1 = 0;
    for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x) {
        uint32_t key = tk_key22(v[i]);
        c0 += (key > T22);
        c1 += (key == T22);
    }
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        c0 += __shfl_down(c0, off, WAVE);
        c1 += __shfl_down(c1, off, WAVE);
    }
    __shared__ int w0[TK_BLOCK / WAVE], w1[TK_BLOCK / WAVE];
    int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    if (lane == 0) { w0[wid] = c0; w1[wid] = c1; }
    __syncthreads();
    if (threadIdx.x == 0) {
        int t0 = 0, t1 = 0;
        for (int w = 0; w < TK_BLOCK / WAVE; ++w) { t0 += w0[w]; t1 += w1[w]; }
        counts[D[8] + lb] = t0;
        counts[BV + D[8] + lb] = t1;
    }
}

// generic device-side exclusive row scan used by the batched scans
__device__ void bt_scan_row(const int* __restrict__ src, int64_t n,
                            int* __restrict__ dst) {
    int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    __shared__ int wave_tot[QBLOCK / WAVE];
    __shared__ int carry;
    if (threadIdx.x == 0) carry = 0;
    __syncthreads();
    for (int64_t i0 = 0; i0 < n; i0 += blockDim.x) {
        int64_t i = i0 + threadIdx.x;
        int v = (i < n) ? src[i] : 0;
        int incl = v;
        for (int off = 1; off < WAVE; off <<= 1) {
            int up = __shfl_up(incl, off, WAVE);
            if (lane >= off) incl += up;
        }
        if (lane == WAVE - 1) wave_tot[wid] = incl;
        __syncthreads();
        int wbase = 0;
        for (int w = 0; w < wid; ++w) wbase += wave_tot[w];
        if (i < n) dst[i] = carry + wbase + incl - v;
        __syncthreads();
        if (threadIdx.x == 0) {
            int tsum = 0;
            for (int w = 0; w < QBLOCK / WAVE; ++w) tsum += wave_tot[w];
            carry += tsum;
        }
        __syncthreads();
    }
}

// grid = planes*T blocks: scan tensor t's slot row within each plane
__global__ void bt_scan_kernel(const int* __restrict__ counts,
                               const int64_t* __restrict__ desc, int nT,
                               int64_t BV, int* __restrict__ offs) {
    const int t = blockIdx.x % nT;
    const int64_t plane = blockIdx.x / nT;
    const int64_t* D = bt_row(desc, t);
    const int64_t nb = (D[0] + BT_CHUNK - 1) / BT_CHUNK;
    bt_scan_row(counts + plane * BV + D[8], nb, offs + plane * BV + D[8]);
}

// ordered two-class compaction -> out_i only (values are re-read FP-aware
// from the dense tensor at the bloom-recovered positions later)
__global__ void bt_scatter_kernel(const float* __restrict__ vals,
                                  const int64_t* __restrict__ desc,
                                  const int* __restrict__ b2t,
                                  const int* __restrict__ sc,
                                  const int* __restrict__ offs, int64_t BV,
                                  int64_t* __restrict__ out_i,
                                  float* __restrict__ out_v /*nullable*/) {
    const int t = b2t[blockIdx.x];
    const int64_t* D = bt_row(desc, t);
    const int64_t lb = blockIdx.x - D[10];
    const int64_t start = lb * BT_CHUNK;
    const int64_t end = min(start + BT_CHUNK, D[0]);
    const float* __restrict__ v = vals + D[1];
    const uint32_t T22 = (uint32_t)sc[t * 4 + 2];
    const int64_t above = sc[t * 4 + 3];
    const int64_t k = D[2];
    const int64_t need = k - above;
    int64_t* __restrict__ out = out_i + D[3];
    float* __restrict__ ov = out_v ? out_v + D[3] : nullptr;
    int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    __shared__ int wc0[TK_BLOCK / WAVE], wc1[TK_BLOCK / WAVE];
    __shared__ int base0_s, base1_s;
    if (threadIdx.x == 0) {
        base0_s = offs[D[8] + lb];
        base1_s = offs[BV + D[8] + lb];
    }
    __syncthreads();
    for (int64_t i0 = start; i0 < end; i0 += blockDim.x) {
        int64_t i = i0 + threadIdx.x;
        bool in_r = (i < end);
        uint32_t key = in_r ? tk_key22(v[i]) : 0u;
        bool p0 = in_r && (key > T22);
        bool p1 = in_r && (key == T22);
        uint64_t b0 = __ballot(p0), b1 = __ballot(p1);
        if (lane == 0) { wc0[wid] = __popcll(b0); wc1[wid] = __popcll(b1); }
        __syncthreads();
        uint64_t below = (lane == 63) ? (~0ull >> 1) : ((1ull << lane) - 1);
        if (p0) {
            int wbase = 0;
            for (int w = 0; w < wid; ++w) wbase += wc0[w];
            int64_t oi = base0_s + wbase + __popcll(b0 & below);
            out[oi] = i;
            if (ov) ov[oi] = v[i];
        } else if (p1) {
            int wbase = 0;
            for (int w = 0; w < wid; ++w) wbase += wc1[w];
            int64_t ordinal = base1_s + wbase + __popcll(b1 & below);
            if (ordinal < need) {
                out[above + ordinal] = i;
                if (ov) ov[above + ordinal] = v[i];
            }
        }
        __syncthreads();
        if (threadIdx.x == 0) {
            int t0 = 0, t1 = 0;
            for (int w = 0; w < TK_BLOCK / WAVE; ++w) { t0 += wc0[w]; t1 += wc1[w]; }
            base0_s += t0;
            base1_s += t1;
        }
        __syncthreads();
    }
}

// hash every selected index into its tensor's bloom bit array in the wire
__global__ void bt_insert_kernel(const int64_t* __restrict__ out_i,
                                 const int64_t* __restrict__ desc, int nT,
                                 int64_t K, uint8_t* __restrict__ wire) {
    int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; j < K; j += stride) {
        // binary search: largest t with koff[t] <= j
        int lo = 0, hi = nT - 1;
        while (lo < hi) {
            int mid = (lo + hi + 1) >> 1;
            if (bt_row(desc, mid)[3] <= j) lo = mid; else hi = mid - 1;
        }
        const int64_t* D = bt_row(desc, lo);
        uint32_t h1, h2;
        hash_bases(out_i[j], &h1, &h2);
        const int64_t m = D[4];
        uint32_t* bits = (uint32_t*)(wire + D[6]);
        uint32_t x = h1;
        for (int h = 0; h < (int)D[5]; ++h, x += h2) {
            uint64_t pos = bloom_pos(x, m);
            atomicOr(&bits[pos >> 5], 1u << (pos & 31));
        }
    }
}

// universe query over R wire buffers (filters share probe positions):
// counts plane per rank + ballot bit-plane per rank.
//
// Memory path: probes are 32-bit WORD loads (one line
// touch per probe, not a byte gather), and for R == 1 (the compress-side
// query — 43% of compression cycles in profiles/r08) the whole filter is
// staged into LDS once per block when it fits the launch's dynamic share
// (ldsq_words; flagship filters are 10-45 KB): the k probes then hit LDS
// at ~50 cyc instead of L1/L2 at 180+.  Multi-rank decode uses the
// interleaved variant below instead.
__global__ void bt_qcount_kernel(const uint8_t* __restrict__ wires,
                                 int64_t wstride, int R,
                                 const int64_t* __restrict__ desc,
                                 const int* __restrict__ b2t, int64_t BV,
                                 int64_t MW, int ldsq_words, int qhead,
                                 int* __restrict__ qcounts,
                                 uint64_t* __restrict__ mask) {
    extern __shared__ uint32_t ldsq[];
    const int t = b2t[blockIdx.x];
    const int64_t* D = bt_row(desc, t);
    const int64_t lb = blockIdx.x - D[10];
    const int64_t start = lb * BT_CHUNK;
    const int64_t end = min(start + BT_CHUNK, D[0]);
    const int64_t m = D[4];
    const int nh = (int)D[5];
    const int64_t bitoff = D[6];
    const int64_t mwoff = D[9];
    const int64_t mw = (m + 31) >> 5;
    const uint32_t* __restrict__ w32 = (const uint32_t*)(wires + bitoff);
    const bool use_lds = (R == 1) && (mw <= (int64_t)ldsq_words);
    if (use_lds) {
        for (int64_t w = threadIdx.x; w < mw; w += blockDim.x)
            ldsq[w] = w32[w];
        __syncthreads();
    }
    int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    int cnt[MAXR];
    for (int r = 0; r < R; ++r) cnt[r] = 0;
    for (int64_t i0 = start; i0 < end; i0 += blockDim.x) {
        int64_t i = i0 + threadIdx.x;
        unsigned alive = (i < end) ? ((1u << R) - 1) : 0u;
        if (alive) {
            uint32_t h1, h2;
            hash_bases(i, &h1, &h2);
            uint32_t x = h1;
            if (use_lds) {
                for (int h = 0; h < nh && alive; ++h, x += h2) {
                    const uint64_t pos = bloom_pos(x, m);
                    if (!(ldsq[pos >> 5] & (1u << (pos & 31)))) alive = 0u;
                }
            } else {
                // Latency shape (PMC r2e: WAIT:BUSY was 30:1 on the serial
                // early-exit loop): probe the first `head` positions with
                // INDEPENDENT loads issued before any test — one memory
                // round-trip instead of a dependent chain — then finish the
                // few survivors (fill=0.5 => ~6% past 4 probes) serially.
                const int head = nh < qhead ? nh : qhead;
                uint64_t p[4];
                for (int j = 0; j < head; ++j, x += h2) p[j] = bloom_pos(x, m);
                for (int r = 0; r < R; ++r) {
                    if (!(alive & (1u << r))) continue;
                    const uint32_t* __restrict__ w32r =
                        (const uint32_t*)(wires + (int64_t)r * wstride + bitoff);
                    bool ok = true;
                    uint32_t w[4];
                    for (int j = 0; j < head; ++j) w[j] = w32r[p[j] >> 5];
                    for (int j = 0; j < head; ++j)
                        ok &= (w[j] >> (p[j] & 31)) & 1;
                    if (!ok) alive &= ~(1u << r);
                }
                for (int h = head; h < nh && alive; ++h, x += h2) {
                    const uint64_t pos = bloom_pos(x, m);
                    const int64_t word = pos >> 5;
                    const uint32_t bit = 1u << (pos & 31);
                    for (int r = 0; r < R; ++r)
                        if (alive & (1u << r))
                            if (!(((const uint32_t*)(wires + (int64_t)r * wstride
                                                     + bitoff))[word] & bit))
                                alive &= ~(1u << r);
                }
            }
        }
        for (int r = 0; r < R; ++r) {
            uint64_t ball = __ballot(alive & (1u << r));
            cnt[r] += __popcll(ball);
            if (lane == 0)
                mask[r * MW + mwoff + ((i0 + (int64_t)wid * WAVE) >> 6)] = ball;
        }
    }
    __shared__ int wsum[MAXR][QBLOCK / WAVE];
    if (lane == 0)
        for (int r = 0; r < R; ++r) wsum[r][wid] = cnt[r];
    __syncthreads();
    if (threadIdx.x == 0) {
        for (int r = 0; r < R; ++r) {
            int total = 0;
            for (int w = 0; w < QBLOCK / WAVE; ++w) total += wsum[r][w];
            qcounts[r * BV + D[8] + lb] = total;
        }
    }
}

// Interleave the R ranks' bloom bit arrays word-by-word:
//   il[(iloff_t + w) * R + r] = word w of tensor t's filter in rank r's wire
// so the decode-side probe of position pos touches R CONSECUTIVE words
// (one or two cache lines) instead of R lines wstride apart.  One cheap
// streaming pass (R * sum(mw) words) before the universe query.
__global__ void bt_interleave_kernel(const uint8_t* __restrict__ wires,
                                     int64_t wstride, int R,
                                     const int64_t* __restrict__ desc, int nT,
                                     int64_t total_mw,
                                     uint32_t* __restrict__ il) {
    int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const int64_t total = total_mw * R;
    for (; g < total; g += stride) {
        const int64_t widx = g / R;
        const int r = (int)(g - widx * R);
        int lo = 0, hi = nT - 1;   // largest t with iloff[t] <= widx
        while (lo < hi) {
            int mid = (lo + hi + 1) >> 1;
            if (bt_row(desc, mid)[15] <= widx) lo = mid; else hi = mid - 1;
        }
        const int64_t* D = bt_row(desc, lo);
        const uint32_t* w32 = (const uint32_t*)(wires + (int64_t)r * wstride + D[6]);
        il[g] = w32[widx - D[15]];
    }
}

// Multi-rank universe query against the interleaved filter block.
__global__ void bt_qcount_inter_kernel(const uint32_t* __restrict__ il, int R,
                                       const int64_t* __restrict__ desc,
                                       const int* __restrict__ b2t, int64_t BV,
                                       int64_t MW, int qhead,
                                       int* __restrict__ qcounts,
                                       uint64_t* __restrict__ mask) {
    const int t = b2t[blockIdx.x];
    const int64_t* D = bt_row(desc, t);
    const int64_t lb = blockIdx.x - D[10];
    const int64_t start = lb * BT_CHUNK;
    const int64_t end = min(start + BT_CHUNK, D[0]);
    const int64_t m = D[4];
    const int nh = (int)D[5];
    const int64_t mwoff = D[9];
    const uint32_t* __restrict__ ilt = il + D[15] * R;
    int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    int cnt[MAXR];
    for (int r = 0; r < R; ++r) cnt[r] = 0;
    for (int64_t i0 = start; i0 < end; i0 += blockDim.x) {
        int64_t i = i0 + threadIdx.x;
        unsigned alive = (i < end) ? ((1u << R) - 1) : 0u;
        if (alive) {
            uint32_t h1, h2;
            hash_bases(i, &h1, &h2);
            uint32_t x = h1;
            // same independent-head shape as bt_qcount (latency-bound loop)
            const int head = nh < qhead ? nh : qhead;
            uint64_t p[4];
            for (int j = 0; j < head; ++j, x += h2) p[j] = bloom_pos(x, m);
            unsigned drop = 0u;
            for (int j = 0; j < head; ++j) {
                const uint32_t bit = 1u << (p[j] & 31);
                const uint32_t* __restrict__ row = ilt + (p[j] >> 5) * R;
                for (int r = 0; r < R; ++r)
                    if (!(row[r] & bit)) drop |= (1u << r);
            }
            alive &= ~drop;
            for (int h = head; h < nh && alive; ++h, x += h2) {
                const uint64_t pos = bloom_pos(x, m);
                const uint32_t bit = 1u << (pos & 31);
                const uint32_t* __restrict__ row = ilt + (pos >> 5) * R;
                for (int r = 0; r < R; ++r)
                    if (alive & (1u << r))
                        if (!(row[r] & bit)) alive &= ~(1u << r);
            }
        }
        for (int r = 0; r < R; ++r) {
            uint64_t ball = __ballot(alive & (1u << r));
            cnt[r] += __popcll(ball);
            if (lane == 0)
                mask[r * MW + mwoff + ((i0 + (int64_t)wid * WAVE) >> 6)] = ball;
        }
    }
    __shared__ int wsum[MAXR][QBLOCK / WAVE];
    if (lane == 0)
        for (int r = 0; r < R; ++r) wsum[r][wid] = cnt[r];
    __syncthreads();
    if (threadIdx.x == 0) {
        for (int r = 0; r < R; ++r) {
            int total = 0;
            for (int w = 0; w < QBLOCK / WAVE; ++w) total += wsum[r][w];
            qcounts[r * BV + D[8] + lb] = total;
        }
    }
}

// compress-side compaction (R=1): leftmost-k indices -> out_idx, AND the
// FP-aware value gather fused in: wire values = dense values at the
// positions decompress will re-derive
__global__ void bt_qscatter_own_kernel(const uint64_t* __restrict__ mask,
                                       const int* __restrict__ qoffs,
                                       const float* __restrict__ vals /*nullable*/,
                                       const int64_t* __restrict__ desc,
                                       const int* __restrict__ b2t,
                                       uint8_t* __restrict__ wire /*nullable*/,
                                       float* __restrict__ vals_out /*nullable*/,
                                       int wire_half,
                                       int64_t* __restrict__ out_idx) {
    const int t = b2t[blockIdx.x];
    const int64_t* D = bt_row(desc, t);
    const int64_t lb = blockIdx.x - D[10];
    const int64_t start = lb * BT_CHUNK;
    const int64_t end = min(start + BT_CHUNK, D[0]);
    const int64_t k = D[2];
    const int64_t mwoff = D[9];
    const float* __restrict__ v = vals ? vals + D[1] : nullptr;
    float* __restrict__ wv = wire ? (float*)(wire + D[7]) : nullptr;
    __half* __restrict__ wh = wire ? (__half*)(wire + D[7]) : nullptr;
    float* __restrict__ vo = vals_out ? vals_out + D[3] : nullptr;
    int64_t* __restrict__ oi = out_idx + D[3];
    int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    __shared__ int wave_cnt[QBLOCK / WAVE];
    __shared__ int base_s;
    if (threadIdx.x == 0) base_s = qoffs[D[8] + lb];
    __syncthreads();
    for (int64_t i0 = start; i0 < end; i0 += blockDim.x) {
        int64_t i = i0 + threadIdx.x;
        uint64_t ball = mask[mwoff + ((i0 + (int64_t)wid * WAVE) >> 6)];
        bool pred = (ball >> lane) & 1;
        if (lane == 0) wave_cnt[wid] = __popcll(ball);
        __syncthreads();
        if (pred) {
            int wbase = 0;
            for (int w = 0; w < wid; ++w) wbase += wave_cnt[w];
            uint64_t below = (lane == 63) ? (~0ull >> 1) : ((1ull << lane) - 1);
            int64_t ord = base_s + wbase + __popcll(ball & below);
            if (ord < k) {
                oi[ord] = i;
                if (wv) {
                    if (wire_half) wh[ord] = __float2half(v[i]);
                    else wv[ord] = v[i];
                }
                if (vo) vo[ord] = v[i];
            }
        }
        __syncthreads();
        if (threadIdx.x == 0) {
            int tsum = 0;
            for (int w = 0; w < QBLOCK / WAVE; ++w) tsum += wave_cnt[w];
            base_s += tsum;
        }
        __syncthreads();
    }
}

// Fused multi-rank decode compaction: ONE candidate-parallel launch
// replaces R sequential bt_qscatter_add launches.  Each thread owns one
// universe candidate, walks the R ballot planes, accumulates every
// rank's wire value for that candidate IN REGISTER, and writes dense[i]
// once — R read-modify-write passes over the dense buffer become one,
// and the accumulation order (r ascending within one thread) is
// deterministic and identical on every rank.
__global__ void bt_qscatter_add_fused_kernel(const uint64_t* __restrict__ mask,
                                             const int* __restrict__ qoffs,
                                             int R, int64_t BV, int64_t MW,
                                             const uint8_t* __restrict__ wires,
                                             int64_t wstride, int wire_half,
                                             const int64_t* __restrict__ desc,
                                             const int* __restrict__ b2t,
                                             float* __restrict__ dense) {
    const int t = b2t[blockIdx.x];
    const int64_t* D = bt_row(desc, t);
    const int64_t lb = blockIdx.x - D[10];
    const int64_t start = lb * BT_CHUNK;
    const int64_t end = min(start + BT_CHUNK, D[0]);
    const int64_t k = D[2];
    const int64_t mwoff = D[9];
    float* __restrict__ dv = dense + D[1];
    const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    __shared__ int wave_cnt[MAXR][QBLOCK / WAVE];
    __shared__ int base_s[MAXR];
    if (threadIdx.x < R)
        base_s[threadIdx.x] = qoffs[(int64_t)threadIdx.x * BV + D[8] + lb];
    __syncthreads();
    for (int64_t i0 = start; i0 < end; i0 += blockDim.x) {
        const int64_t i = i0 + threadIdx.x;
        const int64_t mslot = mwoff + ((i0 + (int64_t)wid * WAVE) >> 6);
        if (lane == 0)
            for (int r = 0; r < R; ++r)
                wave_cnt[r][wid] = __popcll(mask[(int64_t)r * MW + mslot]);
        __syncthreads();
        float acc = 0.0f;
        bool any = false;
        for (int r = 0; r < R; ++r) {
            const uint64_t ball = mask[(int64_t)r * MW + mslot];
            if ((ball >> lane) & 1) {
                int wbase = 0;
                for (int w = 0; w < wid; ++w) wbase += wave_cnt[r][w];
                const uint64_t below = (lane == 63) ? (~0ull >> 1)
                                                    : ((1ull << lane) - 1);
                const int64_t ord = base_s[r] + wbase + __popcll(ball & below);
                if (ord < k) {
                    const uint8_t* wp = wires + (int64_t)r * wstride + D[7];
                    acc += wire_half
                               ? __half2float(((const __half*)wp)[ord])
                               : ((const float*)wp)[ord];
                    any = true;
                }
            }
        }
        __syncthreads();
        if (threadIdx.x < R) {
            int tsum = 0;
            for (int w = 0; w < QBLOCK / WAVE; ++w)
                tsum += wave_cnt[threadIdx.x][w];
            base_s[threadIdx.x] += tsum;
        }
        __syncthreads();
        if (any) dv[i] += acc;
    }
}

// own-payload decode: dense[voff + out_idx] = wire values (indices unique)
__global__ void bt_scatter_dense_kernel(const int64_t* __restrict__ out_idx,
                                        const uint8_t* __restrict__ wire,
                                        const int64_t* __restrict__ desc,
                                        int nT, int64_t K, int wire_half,
                                        float* __restrict__ dense) {
    int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; j < K; j += stride) {
        int lo = 0, hi = nT - 1;
        while (lo < hi) {
            int mid = (lo + hi + 1) >> 1;
            if (bt_row(desc, mid)[3] <= j) lo = mid; else hi = mid - 1;
        }
        const int64_t* D = bt_row(desc, lo);
        const int64_t local = j - D[3];
        dense[D[1] + out_idx[j]] = wire_half
            ? __half2float(((const __half*)(wire + D[7]))[local])
            : ((const float*)(wire + D[7]))[local];
    }
}

__global__ void bt_fill_zero_f(float* __restrict__ p, int64_t n) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) p[i] = 0.f;
}

// ---------------------------------------------------------------------------
// batched drivers
// ---------------------------------------------------------------------------

static inline int bt_grid(int64_t n) {
    return (int)std::min<int64_t>(ceil_div(n, 256), 4096);
}

// Compress all tensors: flat values -> (wire, out_idx).  Every size below is
// host-static for a fixed model layout, so the whole call is sync-free and
// hipGraph-capturable.
std::vector<torch::Tensor> batched_compress(torch::Tensor values_flat,
                                            torch::Tensor desc,
                                            torch::Tensor b2t,
                                            int64_t wire_bytes, int64_t k_total,
                                            int64_t mask_words,
                                            int64_t wire_half,
                                            int64_t ldsq_bytes) {
    CHECK_CUDA(values_flat);
    auto v = values_flat.contiguous();
    auto d = desc.contiguous();
    auto map = b2t.contiguous();
    const int T = (int)d.size(0);
    const int64_t BV = map.numel();
    auto dev = v.device();
    hipStream_t stream = at::hip::getCurrentHIPStream();

    auto ws = torch::empty({(int64_t)T * (2 * TK_BINS + 4) + 6 * BV},
                           torch::dtype(torch::kInt32).device(dev));
    int* hist1 = ws.data_ptr<int>();
    int* hist2 = hist1 + (int64_t)T * TK_BINS;
    int* sc = hist2 + (int64_t)T * TK_BINS;
    int* counts = sc + (int64_t)T * 4;   // 2 planes
    int* offs = counts + 2 * BV;         // 2 planes
    int* qcounts = offs + 2 * BV;
    int* qoffs = qcounts + BV;
    auto mask = torch::empty({mask_words}, torch::dtype(torch::kInt64).device(dev));
    auto wire = torch::empty({wire_bytes}, torch::dtype(torch::kUInt8).device(dev));
    auto out_idx = torch::empty({k_total}, torch::dtype(torch::kInt64).device(dev));

    zero_ints(hist1, (int64_t)T * (2 * TK_BINS + 4), stream);
    zero_ints((int*)wire.data_ptr<uint8_t>(), wire_bytes / 4, stream);  // wire_bytes % 8 == 0

    const int64_t* dp = d.data_ptr<int64_t>();
    const int* mp = map.data_ptr<int>();
    const float* vp = v.data_ptr<float>();

    hipLaunchKernelGGL(bt_hist_kernel, dim3((int)BV), dim3(TK_BLOCK), 0, stream,
                       vp, dp, mp, 1, hist1, sc);
    hipLaunchKernelGGL(bt_thresh_kernel, dim3(T), dim3(WAVE), 0, stream, hist1, dp, 1, sc);
    hipLaunchKernelGGL(bt_hist_kernel, dim3((int)BV), dim3(TK_BLOCK), 0, stream,
                       vp, dp, mp, 2, hist2, sc);
    hipLaunchKernelGGL(bt_thresh_kernel, dim3(T), dim3(WAVE), 0, stream, hist2, dp, 2, sc);
    hipLaunchKernelGGL(bt_count_kernel, dim3((int)BV), dim3(TK_BLOCK), 0, stream,
                       vp, dp, mp, sc, BV, counts);
    hipLaunchKernelGGL(bt_scan_kernel, dim3(2 * T), dim3(QBLOCK), 0, stream,
                       counts, dp, T, BV, offs);
    hipLaunchKernelGGL(bt_scatter_kernel, dim3((int)BV), dim3(TK_BLOCK), 0, stream,
                       vp, dp, mp, sc, offs, BV, out_idx.data_ptr<int64_t>(),
                       (float*)nullptr);
    hipLaunchKernelGGL(bt_insert_kernel, dim3(bt_grid(k_total)), dim3(256), 0, stream,
                       out_idx.data_ptr<int64_t>(), dp, T, k_total,
                       wire.data_ptr<uint8_t>());
    hipLaunchKernelGGL(bt_qcount_kernel, dim3((int)BV), dim3(QBLOCK),
                       (size_t)ldsq_bytes, stream,
                       wire.data_ptr<uint8_t>(), wire_bytes, 1, dp, mp, BV,
                       mask_words, (int)(ldsq_bytes / 4), qhead_env(), qcounts,
                       (uint64_t*)mask.data_ptr<int64_t>());
    hipLaunchKernelGGL(bt_scan_kernel, dim3(T), dim3(QBLOCK), 0, stream,
                       qcounts, dp, T, BV, qoffs);
    hipLaunchKernelGGL(bt_qscatter_own_kernel, dim3((int)BV), dim3(QBLOCK), 0, stream,
                       (const uint64_t*)mask.data_ptr<int64_t>(), qoffs, vp, dp, mp,
                       wire.data_ptr<uint8_t>(), (float*)nullptr, (int)wire_half,
                       out_idx.data_ptr<int64_t>());
    return {wire, out_idx};
}

// Own-payload decode: (wire, out_idx) -> dense flat
torch::Tensor batched_scatter_dense(torch::Tensor wire, torch::Tensor out_idx,
                                    torch::Tensor desc, int64_t total_values,
                                    int64_t wire_half) {
    CHECK_CUDA(wire);
    auto d = desc.contiguous();
    const int T = (int)d.size(0);
    const int64_t K = out_idx.numel();
    auto dense = torch::empty({total_values},
                              torch::dtype(torch::kFloat32).device(wire.device()));
    hipStream_t stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(bt_fill_zero_f, dim3(bt_grid(total_values)), dim3(256), 0, stream,
                       dense.data_ptr<float>(), total_values);
    hipLaunchKernelGGL(bt_scatter_dense_kernel, dim3(bt_grid(K)), dim3(256), 0, stream,
                       out_idx.data_ptr<int64_t>(), wire.data_ptr<uint8_t>(),
                       d.data_ptr<int64_t>(), T, K, (int)wire_half,
                       dense.data_ptr<float>());
    return dense;
}

// Multi-rank decode: stacked wires [R, W] -> SUM of dense decodes.
// Hashing runs once for all R filters; per-rank scatter-adds are launched
// sequentially so the accumulation order is deterministic on every rank.
torch::Tensor batched_decode_sum(torch::Tensor wires2d, torch::Tensor desc,
                                 torch::Tensor b2t, int64_t total_values,
                                 int64_t mask_words, int64_t wire_half,
                                 int64_t total_mw, int64_t ldsq_bytes) {
    CHECK_CUDA(wires2d);
    TORCH_CHECK(wires2d.dim() == 2, "expected [R, W]");
    auto w = wires2d.contiguous();
    auto d = desc.contiguous();
    auto map = b2t.contiguous();
    const int R = (int)w.size(0);
    TORCH_CHECK(R >= 1 && R <= MAXR, "1..16 ranks supported");
    const int64_t W = w.size(1);
    const int T = (int)d.size(0);
    const int64_t BV = map.numel();
    auto dev = w.device();
    hipStream_t stream = at::hip::getCurrentHIPStream();

    auto ws = torch::empty({2 * (int64_t)R * BV}, torch::dtype(torch::kInt32).device(dev));
    int* qcounts = ws.data_ptr<int>();
    int* qoffs = qcounts + (int64_t)R * BV;
    auto mask = torch::empty({(int64_t)R * mask_words},
                             torch::dtype(torch::kInt64).device(dev));
    auto dense = torch::empty({total_values}, torch::dtype(torch::kFloat32).device(dev));
    hipLaunchKernelGGL(bt_fill_zero_f, dim3(bt_grid(total_values)), dim3(256), 0, stream,
                       dense.data_ptr<float>(), total_values);
    if (R > 1 && total_mw > 0) {
        auto il = torch::empty({(int64_t)R * total_mw},
                               torch::dtype(torch::kInt32).device(dev));
        hipLaunchKernelGGL(bt_interleave_kernel, dim3(bt_grid(total_mw * R)),
                           dim3(256), 0, stream, w.data_ptr<uint8_t>(), W, R,
                           d.data_ptr<int64_t>(), T, total_mw,
                           (uint32_t*)il.data_ptr<int>());
        hipLaunchKernelGGL(bt_qcount_inter_kernel, dim3((int)BV), dim3(QBLOCK), 0,
                           stream, (const uint32_t*)il.data_ptr<int>(), R,
                           d.data_ptr<int64_t>(), map.data_ptr<int>(), BV,
                           mask_words, qhead_env(), qcounts,
                           (uint64_t*)mask.data_ptr<int64_t>());
    } else {
        hipLaunchKernelGGL(bt_qcount_kernel, dim3((int)BV), dim3(QBLOCK),
                           (size_t)ldsq_bytes, stream,
                           w.data_ptr<uint8_t>(), W, R, d.data_ptr<int64_t>(),
                           map.data_ptr<int>(), BV, mask_words,
                           (int)(ldsq_bytes / 4), qhead_env(), qcounts,
                           (uint64_t*)mask.data_ptr<int64_t>());
    }
    hipLaunchKernelGGL(bt_scan_kernel, dim3(R * T), dim3(QBLOCK), 0, stream,
                       qcounts, d.data_ptr<int64_t>(), T, BV, qoffs);
    hipLaunchKernelGGL(bt_qscatter_add_fused_kernel, dim3((int)BV), dim3(QBLOCK),
                       0, stream, (const uint64_t*)mask.data_ptr<int64_t>(),
                       qoffs, R, BV, mask_words, w.data_ptr<uint8_t>(), W,
                       (int)wire_half, d.data_ptr<int64_t>(),
                       map.data_ptr<int>(), dense.data_ptr<float>());
    return dense;
}


// ---------------------------------------------------------------------------
// Fused piecewise-polynomial fit (PolyFit/PolySeg value codecs).
//
// The torch path built an [N, 2d+1] float64 power matrix and reduced it
// with index_add_ — fp64 atomics onto ~20 rows measured 760 us PER CALL
// (profiles/NOTES.md r05).  Here: one block per segment accumulates the
// power sums S_p = sum x^p and moments M_p = sum x^p*y (x = (i+1)/len,
// the normalized abscissa) in registers, reduces through LDS, builds the
// ridged Gram system and solves it in-register (Cholesky, d1 <= 8) —
// one kernel for the whole fit, one for the eval.
// ---------------------------------------------------------------------------

#define PF_MAXD1 8  // degree <= 7

__global__ void polyfit_fit_kernel(const float* __restrict__ y,
                                   const int64_t* __restrict__ seg_starts /*[S+1]*/,
                                   int degree, double* __restrict__ coeffs /*[S,d1]*/) {
    const int s = blockIdx.x;
    const int64_t start = seg_starts[s];
    const int64_t end = seg_starts[s + 1];
    const int64_t len = end - start;
    const int d1 = degree + 1;
    const int np = 2 * degree + 1;
    const double inv_len = len > 0 ? 1.0 / (double)len : 1.0;

    double ps[2 * PF_MAXD1 - 1];
    double mo[PF_MAXD1];
    for (int p = 0; p < np; ++p) ps[p] = 0.0;
    for (int p = 0; p < d1; ++p) mo[p] = 0.0;

    for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x) {
        const double x = (double)(i - start + 1) * inv_len;
        const double yv = (double)y[i];
        double xp = 1.0;
        for (int p = 0; p < np; ++p) {
            ps[p] += xp;
            if (p < d1) mo[p] += xp * yv;
            xp *= x;
        }
    }
    // wave reduce, then cross-wave through LDS
    const int lane = threadIdx.x % WAVE, wid = threadIdx.x / WAVE;
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        for (int p = 0; p < np; ++p) ps[p] += __shfl_down(ps[p], off, WAVE);
        for (int p = 0; p < d1; ++p) mo[p] += __shfl_down(mo[p], off, WAVE);
    }
    __shared__ double sh[QBLOCK / WAVE][2 * PF_MAXD1 - 1 + PF_MAXD1];
    if (lane == 0) {
        for (int p = 0; p < np; ++p) sh[wid][p] = ps[p];
        for (int p = 0; p < d1; ++p) sh[wid][np + p] = mo[p];
    }
    __syncthreads();
    if (threadIdx.x != 0) return;
    for (int w = 1; w < (int)(blockDim.x / WAVE); ++w) {
        for (int p = 0; p < np; ++p) ps[p] += sh[w][p];
        for (int p = 0; p < d1; ++p) mo[p] += sh[w][np + p];
    }
    // gram[i][j] = S_{i+j}; ridge = max|diag| * 1e-10 + 1e-30
    double G[PF_MAXD1][PF_MAXD1];
    double dmax = 0.0;
    for (int i = 0; i < d1; ++i) {
        double dg = fabs(ps[2 * i]);
        if (dg > dmax) dmax = dg;
    }
    const double ridge = dmax * 1e-10 + 1e-30;
    for (int i = 0; i < d1; ++i)
        for (int j = 0; j < d1; ++j)
            G[i][j] = ps[i + j] + (i == j ? ridge : 0.0);
    // in-register Cholesky solve G c = mo
    double L[PF_MAXD1][PF_MAXD1];
    for (int i = 0; i < d1; ++i) {
        for (int j = 0; j <= i; ++j) {
            double sum = G[i][j];
            for (int p = 0; p < j; ++p) sum -= L[i][p] * L[j][p];
            if (i == j)
                L[i][j] = sqrt(sum > 1e-300 ? sum : 1e-300);
            else
                L[i][j] = sum / L[j][j];
        }
    }
    double yv[PF_MAXD1];
    for (int i = 0; i < d1; ++i) {
        double sum = mo[i];
        for (int p = 0; p < i; ++p) sum -= L[i][p] * yv[p];
        yv[i] = sum / L[i][i];
    }
    double* c = coeffs + (int64_t)s * d1;
    for (int i = d1 - 1; i >= 0; --i) {
        double sum = yv[i];
        for (int p = i + 1; p < d1; ++p) sum -= L[p][i] * c[p];
        c[i] = sum / L[i][i];
    }
}

// Horner evaluation of the fitted piecewise polynomial (binary search for
// the segment; S is tiny so the search stays in cache)
__global__ void polyfit_eval_kernel(const double* __restrict__ coeffs,
                                    const int64_t* __restrict__ seg_starts,
                                    int S, int degree, int64_t N,
                                    float* __restrict__ out) {
    const int d1 = degree + 1;
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < N; i += stride) {
        int lo = 0, hi = S - 1;
        while (lo < hi) {
            int mid = (lo + hi + 1) >> 1;
            if (seg_starts[mid] <= i) lo = mid; else hi = mid - 1;
        }
        const int64_t start = seg_starts[lo];
        const int64_t len = seg_starts[lo + 1] - start;
        const double x = (double)(i - start + 1) / (double)(len > 0 ? len : 1);
        const double* c = coeffs + (int64_t)lo * d1;
        double yv = c[d1 - 1];
        for (int p = d1 - 2; p >= 0; --p) yv = yv * x + c[p];
        out[i] = (float)yv;
    }
}


// Segment boundaries of the padded 22-slot polyfit layout, derived
// ON-DEVICE from num_pos (codecs/polyfit.py get_segments math) — keeps the
// whole codec sync-free.  ratios = 1/5 .. 1/100000; a slot is zero-length
// when int(num*r) <= 30 (reference condition, pytorch/deepreduce.py:370-374).
__global__ void polyfit_starts_kernel(const double* __restrict__ num_pos_p, int64_t N,
                                      int64_t* __restrict__ starts /*[2*NR+3]*/) {
    if (threadIdx.x != 0 || blockIdx.x != 0) return;
    const double RA[10] = {1.0 / 5, 1.0 / 10, 1.0 / 30, 1.0 / 100, 1.0 / 300,
                           1.0 / 1000, 1.0 / 3000, 1.0 / 10000, 1.0 / 30000,
                           1.0 / 100000};
    const int NR = 10;
    int64_t np = (int64_t)(*num_pos_p + 0.5);
    int64_t nn = N - np;
    // active ratio subset: int(N*r) > 30 (host derives the same count, so
    // the payload slot layout matches codecs/polyfit.py s_pad(N))
    int NA = 0;
    for (int i = 0; i < NR; ++i)
        if ((int64_t)((double)N * RA[i]) > 30) NA = i + 1;
    int64_t pos[10], neg[10], psum = 0, nsum = 0;
    for (int i = 0; i < NA; ++i) {
        int64_t p = (int64_t)((double)np * RA[i]);
        int64_t n = (int64_t)((double)nn * RA[i]);
        pos[i] = (p > 30) ? p : 0;
        neg[i] = (n > 30) ? n : 0;
        psum += pos[i];
        nsum += neg[i];
    }
    int64_t seg[2 * 10 + 2];
    for (int i = 0; i < NA; ++i) seg[i] = pos[NA - 1 - i];  // reversed
    seg[NA] = np - psum;
    seg[NA + 1] = nn - nsum;
    for (int i = 0; i < NA; ++i) seg[NA + 2 + i] = neg[i];
    int64_t acc = 0;
    starts[0] = 0;
    for (int i = 0; i < 2 * NA + 2; ++i) {
        acc += seg[i];
        starts[i + 1] = acc;
    }
}

static inline int64_t polyfit_s_pad(int64_t N) {
    const double RA[10] = {1.0 / 5, 1.0 / 10, 1.0 / 30, 1.0 / 100, 1.0 / 300,
                           1.0 / 1000, 1.0 / 3000, 1.0 / 10000, 1.0 / 30000,
                           1.0 / 100000};
    int64_t na = 0;
    for (int i = 0; i < 10; ++i)
        if ((int64_t)((double)N * RA[i]) > 30) na = i + 1;
    return 2 * na + 2;
}

torch::Tensor polyfit_starts(torch::Tensor num_pos, int64_t N) {
    CHECK_CUDA(num_pos);
    auto npd = num_pos.to(torch::kFloat64).contiguous();
    auto starts = torch::empty({polyfit_s_pad(N) + 1},
                               torch::dtype(torch::kInt64).device(num_pos.device()));
    hipStream_t stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(polyfit_starts_kernel, dim3(1), dim3(WAVE), 0, stream,
                       npd.data_ptr<double>(), N, starts.data_ptr<int64_t>());
    return starts;
}

torch::Tensor polyfit_fit(torch::Tensor y, torch::Tensor seg_starts, int64_t degree) {
    CHECK_CUDA(y);
    TORCH_CHECK(degree + 1 <= PF_MAXD1, "degree <= 7");
    auto yy = y.to(torch::kFloat32).contiguous();
    auto ss = seg_starts.to(torch::kInt64).contiguous();
    const int S = (int)ss.numel() - 1;
    auto coeffs = torch::empty({S, degree + 1},
                               torch::dtype(torch::kFloat64).device(y.device()));
    if (S > 0) {
        hipStream_t stream = at::hip::getCurrentHIPStream();
        hipLaunchKernelGGL(polyfit_fit_kernel, dim3(S), dim3(QBLOCK), 0, stream,
                           yy.data_ptr<float>(), ss.data_ptr<int64_t>(), (int)degree,
                           coeffs.data_ptr<double>());
    }
    return coeffs;
}

torch::Tensor polyfit_eval(torch::Tensor coeffs, torch::Tensor seg_starts, int64_t N) {
    CHECK_CUDA(coeffs);
    auto cc = coeffs.to(torch::kFloat64).contiguous();
    auto ss = seg_starts.to(torch::kInt64).contiguous();
    const int S = (int)ss.numel() - 1;
    const int degree = (int)cc.size(1) - 1;
    auto out = torch::empty({N}, torch::dtype(torch::kFloat32).device(coeffs.device()));
    if (N > 0) {
        hipStream_t stream = at::hip::getCurrentHIPStream();
        hipLaunchKernelGGL(polyfit_eval_kernel, dim3(bt_grid(N)), dim3(256), 0, stream,
                           cc.data_ptr<double>(), ss.data_ptr<int64_t>(), S, degree, N,
                           out.data_ptr<float>());
    }
    return out;
}


// ===========================================================================
// Batched 'both'-mode pipeline (bloom index + polyfit values + packed
// mapping), building on the bt_* machinery.  Wire layout per tensor —
// byte-identical to the generic _flatten_payload of the 'both' wrapper:
//   [coeffs f64 (sp*d1+1), pad8] [bloom bits u8 ceil(m/8), pad8]
//   [mapping u8: 5B header (count LE32, nbits) + ceil(k*nbits/8), pad8]
// Extra descriptor columns (BT2_F layout, cols 0..11 as BT_F):
//  11 sp       padded polyfit slot count (s_pad(k))
//  12 coeffoff byte offset of the f64 coeff block in the wire
//  13 mapoff   byte offset of the mapping block in the wire
//  14 nbits    mapping bit width (= bitlength(k-1))
//  15 kmax     row stride of the padded sort matrix (globally uniform)
// ===========================================================================

#define PF_SMAX 23  // starts row stride (<= 22 slots + 1)
#define bt2_row bt_row

// scatter the bloom-ordered values into the padded [T, kmax] sort matrix
// (pad = -inf so descending sort pushes pads to the tail)
__global__ void bt2_padmat_kernel(const float* __restrict__ vals /*[K]*/,
                                  const int64_t* __restrict__ desc, int nT,
                                  int64_t K, int64_t kmax,
                                  float* __restrict__ padmat) {
    int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; j < K; j += stride) {
        int lo = 0, hi = nT - 1;
        while (lo < hi) {
            int mid = (lo + hi + 1) >> 1;
            if (bt2_row(desc, mid)[3] <= j) lo = mid; else hi = mid - 1;
        }
        padmat[(int64_t)lo * kmax + (j - bt2_row(desc, lo)[3])] = vals[j];
    }
}

__global__ void bt2_fill_ninf(float* __restrict__ p, int64_t n) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) p[i] = -INFINITY;
}

// per-tensor: derive padded segment starts from num_pos (same math as
// polyfit_starts_kernel), write them to starts[T, PF_SMAX], and write the
// num_pos trailer + the 5-byte mapping header into the wire
__global__ void bt2_starts_kernel(const double* __restrict__ num_pos /*[T]*/,
                                  const int64_t* __restrict__ desc,
                                  int64_t* __restrict__ starts /*[T,PF_SMAX]*/,
                                  uint8_t* __restrict__ wire, int d1) {
    const int t = blockIdx.x;
    if (threadIdx.x != 0) return;
    const int64_t* D = bt2_row(desc, t);
    const int64_t N = D[2];  // polyfit N = k (values per tensor)
    const double RA[10] = {1.0 / 5, 1.0 / 10, 1.0 / 30, 1.0 / 100, 1.0 / 300,
                           1.0 / 1000, 1.0 / 3000, 1.0 / 10000, 1.0 / 30000,
                           1.0 / 100000};
    const int NR = 10;
    int NA = 0;
    for (int i = 0; i < NR; ++i)
        if ((int64_t)((double)N * RA[i]) > 30) NA = i + 1;
    int64_t np = (int64_t)(num_pos[t] + 0.5);
    int64_t nn = N - np;
    int64_t pos[10], neg[10], psum = 0, nsum = 0;
    for (int i = 0; i < NA; ++i) {
        int64_t p = (int64_t)((double)np * RA[i]);
        int64_t n = (int64_t)((double)nn * RA[i]);
        pos[i] = (p > 30) ? p : 0;
        neg[i] = (n > 30) ? n : 0;
        psum += pos[i];
        nsum += neg[i];
    }
    int64_t seg[2 * 10 + 2];
    for (int i = 0; i < NA; ++i) seg[i] = pos[NA - 1 - i];
    seg[NA] = np - psum;
    seg[NA + 1] = nn - nsum;
    for (int i = 0; i < NA; ++i) seg[NA + 2 + i] = neg[i];
    int64_t* st = starts + (int64_t)t * PF_SMAX;
    int64_t acc = 0;
    st[0] = 0;
    for (int i = 0; i < 2 * NA + 2; ++i) {
        acc += seg[i];
        st[i + 1] = acc;
    }
    // wire trailer: num_pos after the sp*d1 coefficients
    const int64_t sp = D[11];
    double* cw = (double*)(wire + D[12]);
    cw[sp * d1] = (double)np;
    // mapping header: count LE32 + nbits (both-mode only; value-mode
    // descriptors carry D[13] < 0)
    if (D[13] >= 0) {
        uint8_t* mh = wire + D[13];
        const int64_t k = D[2];
        mh[0] = (uint8_t)(k & 255);
        mh[1] = (uint8_t)((k >> 8) & 255);
        mh[2] = (uint8_t)((k >> 16) & 255);
        mh[3] = (uint8_t)((k >> 24) & 255);
        mh[4] = (uint8_t)D[14];
    }
}

// fit all (tensor, segment) pairs: grid = segmap length; y rows come from
// the sorted pad matrix; coeffs are written straight into the wire
__global__ void bt2_fit_kernel(const float* __restrict__ sorted /*[T,kmax]*/,
                               const int64_t* __restrict__ desc,
                               const int64_t* __restrict__ starts /*[T,PF_SMAX]*/,
                               const int* __restrict__ seg_t,
                               const int* __restrict__ seg_i, int degree,
                               int64_t kmax, uint8_t* __restrict__ wire) {
    const int t = seg_t[blockIdx.x];
    const int si = seg_i[blockIdx.x];
    const int64_t* D = bt2_row(desc, t);
    const int64_t* st = starts + (int64_t)t * PF_SMAX;
    const int64_t start = st[si];
    const int64_t end = st[si + 1];
    const int64_t len = end - start;
    const int d1 = degree + 1;
    const int np = 2 * degree + 1;
    const double inv_len = len > 0 ? 1.0 / (double)len : 1.0;
    const float* __restrict__ y = sorted + (int64_t)t * kmax;

    double ps[2 * PF_MAXD1 - 1];
    double mo[PF_MAXD1];
    for (int p = 0; p < np; ++p) ps[p] = 0.0;
    for (int p = 0; p < d1; ++p) mo[p] = 0.0;
    for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x) {
        const double x = (double)(i - start + 1) * inv_len;
        const double yv = (double)y[i];
        double xp = 1.0;
        for (int p = 0; p < np; ++p) {
            ps[p] += xp;
            if (p < d1) mo[p] += xp * yv;
            xp *= x;
        }
    }
    const int lane = threadIdx.x % WAVE, wid = threadIdx.x / WAVE;
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        for (int p = 0; p < np; ++p) ps[p] += __shfl_down(ps[p], off, WAVE);
        for (int p = 0; p < d1; ++p) mo[p] += __shfl_down(mo[p], off, WAVE);
    }
    __shared__ double sh[QBLOCK / WAVE][2 * PF_MAXD1 - 1 + PF_MAXD1];
    if (lane == 0) {
        for (int p = 0; p < np; ++p) sh[wid][p] = ps[p];
        for (int p = 0; p < d1; ++p) sh[wid][np + p] = mo[p];
    }
    __syncthreads();
    if (threadIdx.x != 0) return;
    for (int w = 1; w < (int)(blockDim.x / WAVE); ++w) {
        for (int p = 0; p < np; ++p) ps[p] += sh[w][p];
        for (int p = 0; p < d1; ++p) mo[p] += sh[w][np + p];
    }
    double G[PF_MAXD1][PF_MAXD1];
    double dmax = 0.0;
    for (int i = 0; i < d1; ++i) {
        double dg = fabs(ps[2 * i]);
        if (dg > dmax) dmax = dg;
    }
    const double ridge = dmax * 1e-10 + 1e-30;
    for (int i = 0; i < d1; ++i)
        for (int j = 0; j < d1; ++j)
            G[i][j] = ps[i + j] + (i == j ? ridge : 0.0);
    double L[PF_MAXD1][PF_MAXD1];
    for (int i = 0; i < d1; ++i) {
        for (int j = 0; j <= i; ++j) {
            double sum = G[i][j];
            for (int p = 0; p < j; ++p) sum -= L[i][p] * L[j][p];
            if (i == j) L[i][j] = sqrt(sum > 1e-300 ? sum : 1e-300);
            else L[i][j] = sum / L[j][j];
        }
    }
    double yv[PF_MAXD1];
    for (int i = 0; i < d1; ++i) {
        double sum = mo[i];
        for (int p = 0; p < i; ++p) sum -= L[i][p] * yv[p];
        yv[i] = sum / L[i][i];
    }
    double* c = (double*)(wire + D[12]) + (int64_t)si * d1;
    double cr[PF_MAXD1];
    for (int i = d1 - 1; i >= 0; --i) {
        double sum = yv[i];
        for (int p = i + 1; p < d1; ++p) sum -= L[p][i] * cr[p];
        cr[i] = sum / L[i][i];
    }
    for (int i = 0; i < d1; ++i) c[i] = cr[i];
}

// pack each tensor's mapping (argsort of the padded sort) at nbits into the
// wire; bit positions can straddle words -> two atomicOr
__global__ void bt2_pack_map_kernel(const int64_t* __restrict__ argsortm /*[T,kmax]*/,
                                    const int64_t* __restrict__ desc, int nT,
                                    int64_t K, int64_t kmax,
                                    uint8_t* __restrict__ wire) {
    int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; j < K; j += stride) {
        int lo = 0, hi = nT - 1;
        while (lo < hi) {
            int mid = (lo + hi + 1) >> 1;
            if (bt2_row(desc, mid)[3] <= j) lo = mid; else hi = mid - 1;
        }
        const int64_t* D = bt2_row(desc, lo);
        const int64_t local = j - D[3];
        const uint64_t v = (uint64_t)argsortm[(int64_t)lo * kmax + local];
        const int nb = (int)D[14];
        // payload starts after the 5-byte header; base bit offset within
        // the word-aligned region starting at mapoff
        const int64_t bit0 = 40 + local * nb;
        uint32_t* w = (uint32_t*)(wire + D[13]);
        const int64_t wi = bit0 >> 5;
        const int sh = (int)(bit0 & 31);
        atomicOr(&w[wi], (uint32_t)(v << sh));
        if (sh + nb > 32) atomicOr(&w[wi + 1], (uint32_t)(v >> (32 - sh)));
    }
}

// evaluate the fitted curves: vals_eval[j] = poly(t, local position)
__global__ void bt2_eval_kernel(const uint8_t* __restrict__ wire_r,
                                const int64_t* __restrict__ desc, int nT,
                                int64_t K,
                                const int64_t* __restrict__ starts /*[T,PF_SMAX]*/,
                                int d1, float* __restrict__ out) {
    int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; j < K; j += stride) {
        int lo = 0, hi = nT - 1;
        while (lo < hi) {
            int mid = (lo + hi + 1) >> 1;
            if (bt2_row(desc, mid)[3] <= j) lo = mid; else hi = mid - 1;
        }
        const int64_t* D = bt2_row(desc, lo);
        const int64_t local = j - D[3];
        const int64_t* st = starts + (int64_t)lo * PF_SMAX;
        const int64_t sp = D[11];
        int s = 0;
        for (int i = 1; i < (int)sp; ++i)
            if (st[i] <= local) s = i;
        const int64_t sstart = st[s];
        const int64_t slen = st[s + 1] - sstart;
        const double x = (double)(local - sstart + 1) / (double)(slen > 0 ? slen : 1);
        const double* c = (const double*)(wire_r + D[12]) + (int64_t)s * d1;
        double yv = c[d1 - 1];
        for (int p = d1 - 2; p >= 0; --p) yv = yv * x + c[p];
        out[j] = (float)yv;
    }
}

// derive starts for a (possibly remote) wire from its transmitted num_pos
__global__ void bt2_starts_from_wire_kernel(const uint8_t* __restrict__ wire_r,
                                            const int64_t* __restrict__ desc,
                                            int d1,
                                            int64_t* __restrict__ starts) {
    const int t = blockIdx.x;
    if (threadIdx.x != 0) return;
    const int64_t* D = bt2_row(desc, t);
    const double np = ((const double*)(wire_r + D[12]))[D[11] * d1];
    const int64_t N = D[2];
    const double RA[10] = {1.0 / 5, 1.0 / 10, 1.0 / 30, 1.0 / 100, 1.0 / 300,
                           1.0 / 1000, 1.0 / 3000, 1.0 / 10000, 1.0 / 30000,
                           1.0 / 100000};
    const int NR = 10;
    int NA = 0;
    for (int i = 0; i < NR; ++i)
        if ((int64_t)((double)N * RA[i]) > 30) NA = i + 1;
    int64_t npi = (int64_t)(np + 0.5);
    int64_t nn = N - npi;
    int64_t pos[10], neg[10], psum = 0, nsum = 0;
    for (int i = 0; i < NA; ++i) {
        int64_t p = (int64_t)((double)npi * RA[i]);
        int64_t n = (int64_t)((double)nn * RA[i]);
        pos[i] = (p > 30) ? p : 0;
        neg[i] = (n > 30) ? n : 0;
        psum += pos[i];
        nsum += neg[i];
    }
    int64_t seg[2 * 10 + 2];
    for (int i = 0; i < NA; ++i) seg[i] = pos[NA - 1 - i];
    seg[NA] = npi - psum;
    seg[NA + 1] = nn - nsum;
    for (int i = 0; i < NA; ++i) seg[NA + 2 + i] = neg[i];
    int64_t* st = starts + (int64_t)t * PF_SMAX;
    int64_t acc = 0;
    st[0] = 0;
    for (int i = 0; i < 2 * NA + 2; ++i) {
        acc += seg[i];
        st[i + 1] = acc;
    }
}

// dense[voff + positives[koff + map(j)]] (+)= vals_eval[koff + j]
// map(j) read straight out of the packed wire bits (no unpack pass)
__global__ void bt2_scatter_both_kernel(const float* __restrict__ vals_eval,
                                        const int64_t* __restrict__ positives,
                                        const uint8_t* __restrict__ wire_r,
                                        const int64_t* __restrict__ desc, int nT,
                                        int64_t K, int accumulate,
                                        float* __restrict__ dense) {
    int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; j < K; j += stride) {
        int lo = 0, hi = nT - 1;
        while (lo < hi) {
            int mid = (lo + hi + 1) >> 1;
            if (bt2_row(desc, mid)[3] <= j) lo = mid; else hi = mid - 1;
        }
        const int64_t* D = bt2_row(desc, lo);
        const int64_t local = j - D[3];
        const int nb = (int)D[14];
        const int64_t bit0 = 40 + local * nb;
        const uint32_t* w = (const uint32_t*)(wire_r + D[13]);
        const int64_t wi = bit0 >> 5;
        const int sh = (int)(bit0 & 31);
        uint64_t acc = (uint64_t)w[wi] >> sh;
        if (sh + nb > 32) acc |= (uint64_t)w[wi + 1] << (32 - sh);
        const int64_t map = (int64_t)(acc & ((nb == 64) ? ~0ull : ((1ull << nb) - 1)));
        const int64_t idx = positives[D[3] + map];
        float* d = dense + D[1] + idx;
        if (accumulate) *d += vals_eval[j];
        else *d = vals_eval[j];
    }
}


// ---------------------------------------------------------------------------
// batched 'both' drivers
// ---------------------------------------------------------------------------

// Compress the whole model in 'both' mode: (wire, own_dense).
std::vector<torch::Tensor> batched_compress_both(
        torch::Tensor values_flat, torch::Tensor desc, torch::Tensor b2t,
        torch::Tensor seg_t, torch::Tensor seg_i, int64_t wire_bytes,
        int64_t k_total, int64_t mask_words, int64_t kmax, int64_t degree,
        int64_t total_values, int64_t ldsq_bytes) {
    CHECK_CUDA(values_flat);
    auto v = values_flat.contiguous();
    auto d = desc.contiguous();
    auto map = b2t.contiguous();
    auto st_t = seg_t.contiguous();
    auto st_i = seg_i.contiguous();
    const int T = (int)d.size(0);
    const int64_t BV = map.numel();
    const int SB = (int)st_t.numel();
    const int d1 = (int)degree + 1;
    auto dev = v.device();
    hipStream_t stream = at::hip::getCurrentHIPStream();

    auto ws = torch::empty({(int64_t)T * (2 * TK_BINS + 4) + 6 * BV},
                           torch::dtype(torch::kInt32).device(dev));
    int* hist1 = ws.data_ptr<int>();
    int* hist2 = hist1 + (int64_t)T * TK_BINS;
    int* sc = hist2 + (int64_t)T * TK_BINS;
    int* counts = sc + (int64_t)T * 4;
    int* offs = counts + 2 * BV;
    int* qcounts = offs + 2 * BV;
    int* qoffs = qcounts + BV;
    auto mask = torch::empty({mask_words}, torch::dtype(torch::kInt64).device(dev));
    auto wire = torch::empty({wire_bytes}, torch::dtype(torch::kUInt8).device(dev));
    auto out_idx = torch::empty({k_total}, torch::dtype(torch::kInt64).device(dev));
    auto vals_tmp = torch::empty({k_total}, torch::dtype(torch::kFloat32).device(dev));
    auto padmat = torch::empty({T, kmax}, torch::dtype(torch::kFloat32).device(dev));
    auto starts = torch::empty({(int64_t)T * PF_SMAX},
                               torch::dtype(torch::kInt64).device(dev));

    zero_ints(hist1, (int64_t)T * (2 * TK_BINS + 4), stream);
    zero_ints((int*)wire.data_ptr<uint8_t>(), wire_bytes / 4, stream);

    const int64_t* dp = d.data_ptr<int64_t>();
    const int* mp = map.data_ptr<int>();
    const float* vp = v.data_ptr<float>();

    hipLaunchKernelGGL(bt_hist_kernel, dim3((int)BV), dim3(TK_BLOCK), 0, stream,
                       vp, dp, mp, 1, hist1, sc);
    hipLaunchKernelGGL(bt_thresh_kernel, dim3(T), dim3(WAVE), 0, stream, hist1, dp, 1, sc);
    hipLaunchKernelGGL(bt_hist_kernel, dim3((int)BV), dim3(TK_BLOCK), 0, stream,
                       vp, dp, mp, 2, hist2, sc);
    hipLaunchKernelGGL(bt_thresh_kernel, dim3(T), dim3(WAVE), 0, stream, hist2, dp, 2, sc);
    hipLaunchKernelGGL(bt_count_kernel, dim3((int)BV), dim3(TK_BLOCK), 0, stream,
                       vp, dp, mp, sc, BV, counts);
    hipLaunchKernelGGL(bt_scan_kernel, dim3(2 * T), dim3(QBLOCK), 0, stream,
                       counts, dp, T, BV, offs);
    hipLaunchKernelGGL(bt_scatter_kernel, dim3((int)BV), dim3(TK_BLOCK), 0, stream,
                       vp, dp, mp, sc, offs, BV, out_idx.data_ptr<int64_t>(),
                       (float*)nullptr);
    hipLaunchKernelGGL(bt_insert_kernel, dim3(bt_grid(k_total)), dim3(256), 0, stream,
                       out_idx.data_ptr<int64_t>(), dp, T, k_total,
                       wire.data_ptr<uint8_t>());
    hipLaunchKernelGGL(bt_qcount_kernel, dim3((int)BV), dim3(QBLOCK),
                       (size_t)ldsq_bytes, stream,
                       wire.data_ptr<uint8_t>(), wire_bytes, 1, dp, mp, BV,
                       mask_words, (int)(ldsq_bytes / 4), qhead_env(), qcounts,
                       (uint64_t*)mask.data_ptr<int64_t>());
    hipLaunchKernelGGL(bt_scan_kernel, dim3(T), dim3(QBLOCK), 0, stream,
                       qcounts, dp, T, BV, qoffs);
    // FP-aware gather to the temp buffer; positives into out_idx
    hipLaunchKernelGGL(bt_qscatter_own_kernel, dim3((int)BV), dim3(QBLOCK), 0, stream,
                       (const uint64_t*)mask.data_ptr<int64_t>(), qoffs, vp, dp, mp,
                       (uint8_t*)nullptr, vals_tmp.data_ptr<float>(), 0,
                       out_idx.data_ptr<int64_t>());
    // padded descending sort (stable: deterministic tie order)
    hipLaunchKernelGGL(bt2_fill_ninf, dim3(bt_grid((int64_t)T * kmax)), dim3(256), 0,
                       stream, padmat.data_ptr<float>(), (int64_t)T * kmax);
    hipLaunchKernelGGL(bt2_padmat_kernel, dim3(bt_grid(k_total)), dim3(256), 0, stream,
                       vals_tmp.data_ptr<float>(), dp, T, k_total, kmax,
                       padmat.data_ptr<float>());
    auto sorted_arg = at::sort(padmat, /*stable=*/true, /*dim=*/1, /*descending=*/true);
    auto sorted = std::get<0>(sorted_arg).contiguous();
    auto argsortm = std::get<1>(sorted_arg).contiguous();
    auto num_pos = (sorted > 0).sum(1).to(torch::kFloat64).contiguous();
    hipLaunchKernelGGL(bt2_starts_kernel, dim3(T), dim3(WAVE), 0, stream,
                       num_pos.data_ptr<double>(), dp,
                       starts.data_ptr<int64_t>(), wire.data_ptr<uint8_t>(), d1);
    hipLaunchKernelGGL(bt2_fit_kernel, dim3(SB), dim3(QBLOCK), 0, stream,
                       sorted.data_ptr<float>(), dp, starts.data_ptr<int64_t>(),
                       st_t.data_ptr<int>(), st_i.data_ptr<int>(), (int)degree,
                       kmax, wire.data_ptr<uint8_t>());
    hipLaunchKernelGGL(bt2_pack_map_kernel, dim3(bt_grid(k_total)), dim3(256), 0, stream,
                       argsortm.data_ptr<int64_t>(), dp, T, k_total, kmax,
                       wire.data_ptr<uint8_t>());
    // own decode: eval the fitted curves and scatter through the mapping
    auto vals_eval = torch::empty({k_total}, torch::dtype(torch::kFloat32).device(dev));
    auto own = torch::empty({total_values}, torch::dtype(torch::kFloat32).device(dev));
    hipLaunchKernelGGL(bt_fill_zero_f, dim3(bt_grid(total_values)), dim3(256), 0, stream,
                       own.data_ptr<float>(), total_values);
    hipLaunchKernelGGL(bt2_eval_kernel, dim3(bt_grid(k_total)), dim3(256), 0, stream,
                       wire.data_ptr<uint8_t>(), dp, T, k_total,
                       starts.data_ptr<int64_t>(), d1, vals_eval.data_ptr<float>());
    hipLaunchKernelGGL(bt2_scatter_both_kernel, dim3(bt_grid(k_total)), dim3(256), 0,
                       stream, vals_eval.data_ptr<float>(),
                       out_idx.data_ptr<int64_t>(), wire.data_ptr<uint8_t>(), dp, T,
                       k_total, 0, own.data_ptr<float>());
    return {wire, own};
}

// Multi-rank 'both' decode: SUM of dense decodes of R stacked wires.
torch::Tensor batched_decode_both_sum(torch::Tensor wires2d, torch::Tensor desc,
                                      torch::Tensor b2t, int64_t total_values,
                                      int64_t mask_words, int64_t k_total,
                                      int64_t degree, int64_t total_mw,
                                      int64_t ldsq_bytes) {
    CHECK_CUDA(wires2d);
    TORCH_CHECK(wires2d.dim() == 2, "expected [R, W]");
    auto w = wires2d.contiguous();
    auto d = desc.contiguous();
    auto map = b2t.contiguous();
    const int R = (int)w.size(0);
    TORCH_CHECK(R >= 1 && R <= MAXR, "1..16 ranks supported");
    const int64_t W = w.size(1);
    const int T = (int)d.size(0);
    const int64_t BV = map.numel();
    const int d1 = (int)degree + 1;
    auto dev = w.device();
    hipStream_t stream = at::hip::getCurrentHIPStream();

    auto ws = torch::empty({2 * (int64_t)R * BV}, torch::dtype(torch::kInt32).device(dev));
    int* qcounts = ws.data_ptr<int>();
    int* qoffs = qcounts + (int64_t)R * BV;
    auto mask = torch::empty({(int64_t)R * mask_words},
                             torch::dtype(torch::kInt64).device(dev));
    auto dense = torch::empty({total_values}, torch::dtype(torch::kFloat32).device(dev));
    auto positives = torch::empty({k_total}, torch::dtype(torch::kInt64).device(dev));
    auto vals_eval = torch::empty({k_total}, torch::dtype(torch::kFloat32).device(dev));
    auto starts = torch::empty({(int64_t)T * PF_SMAX},
                               torch::dtype(torch::kInt64).device(dev));
    hipLaunchKernelGGL(bt_fill_zero_f, dim3(bt_grid(total_values)), dim3(256), 0, stream,
                       dense.data_ptr<float>(), total_values);
    if (R > 1 && total_mw > 0) {
        auto il = torch::empty({(int64_t)R * total_mw},
                               torch::dtype(torch::kInt32).device(dev));
        hipLaunchKernelGGL(bt_interleave_kernel, dim3(bt_grid(total_mw * R)),
                           dim3(256), 0, stream, w.data_ptr<uint8_t>(), W, R,
                           d.data_ptr<int64_t>(), T, total_mw,
                           (uint32_t*)il.data_ptr<int>());
        hipLaunchKernelGGL(bt_qcount_inter_kernel, dim3((int)BV), dim3(QBLOCK), 0,
                           stream, (const uint32_t*)il.data_ptr<int>(), R,
                           d.data_ptr<int64_t>(), map.data_ptr<int>(), BV,
                           mask_words, qhead_env(), qcounts,
                           (uint64_t*)mask.data_ptr<int64_t>());
    } else {
        hipLaunchKernelGGL(bt_qcount_kernel, dim3((int)BV), dim3(QBLOCK),
                           (size_t)ldsq_bytes, stream,
                           w.data_ptr<uint8_t>(), W, R, d.data_ptr<int64_t>(),
                           map.data_ptr<int>(), BV, mask_words,
                           (int)(ldsq_bytes / 4), qhead_env(), qcounts,
                           (uint64_t*)mask.data_ptr<int64_t>());
    }
    hipLaunchKernelGGL(bt_scan_kernel, dim3(R * T), dim3(QBLOCK), 0, stream,
                       qcounts, d.data_ptr<int64_t>(), T, BV, qoffs);
    for (int r = 0; r < R; ++r) {
        const uint8_t* wr = w.data_ptr<uint8_t>() + (int64_t)r * W;
        hipLaunchKernelGGL(bt_qscatter_own_kernel, dim3((int)BV), dim3(QBLOCK), 0, stream,
                           (const uint64_t*)mask.data_ptr<int64_t>() + (int64_t)r * mask_words,
                           qoffs + (int64_t)r * BV, (const float*)nullptr,
                           d.data_ptr<int64_t>(), map.data_ptr<int>(),
                           (uint8_t*)nullptr, (float*)nullptr, 0,
                           positives.data_ptr<int64_t>());
        hipLaunchKernelGGL(bt2_starts_from_wire_kernel, dim3(T), dim3(WAVE), 0, stream,
                           wr, d.data_ptr<int64_t>(), d1, starts.data_ptr<int64_t>());
        hipLaunchKernelGGL(bt2_eval_kernel, dim3(bt_grid(k_total)), dim3(256), 0, stream,
                           wr, d.data_ptr<int64_t>(), T, k_total,
                           starts.data_ptr<int64_t>(), d1, vals_eval.data_ptr<float>());
        hipLaunchKernelGGL(bt2_scatter_both_kernel, dim3(bt_grid(k_total)), dim3(256), 0,
                           stream, vals_eval.data_ptr<float>(),
                           positives.data_ptr<int64_t>(), wr, d.data_ptr<int64_t>(), T,
                           k_total, 1, dense.data_ptr<float>());
    }
    return dense;
}



// ---------------------------------------------------------------------------
// batched 'value' pipeline (polyfit coefficients + raw int32 indices in
// value-sorted order — no bloom).  Wire per tensor, matching the generic
// ValueCompressor payload: [coeffs f64 (sp*d1+1), pad8][idxs int32 k, pad8].
// Descriptor col 6 = idxoff (bytes); col 13 = -1 (no mapping header).
// ---------------------------------------------------------------------------

// wire_idx[ord] = out_idx[koff + argsort(ord)] as int32
__global__ void bt2_write_idx_kernel(const int64_t* __restrict__ out_idx,
                                     const int64_t* __restrict__ argsortm,
                                     const int64_t* __restrict__ desc, int nT,
                                     int64_t K, int64_t kmax,
                                     uint8_t* __restrict__ wire) {
    int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; j < K; j += stride) {
        int lo = 0, hi = nT - 1;
        while (lo < hi) {
            int mid = (lo + hi + 1) >> 1;
            if (bt_row(desc, mid)[3] <= j) lo = mid; else hi = mid - 1;
        }
        const int64_t* D = bt_row(desc, lo);
        const int64_t local = j - D[3];
        const int64_t src = argsortm[(int64_t)lo * kmax + local];
        ((int32_t*)(wire + D[6]))[local] = (int32_t)out_idx[D[3] + src];
    }
}

// dense[voff + wire_idx[ord]] (+)= vals_eval[ord]
__global__ void bt2_scatter_value_kernel(const float* __restrict__ vals_eval,
                                         const uint8_t* __restrict__ wire_r,
                                         const int64_t* __restrict__ desc,
                                         int nT, int64_t K, int accumulate,
                                         float* __restrict__ dense) {
    int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; j < K; j += stride) {
        int lo = 0, hi = nT - 1;
        while (lo < hi) {
            int mid = (lo + hi + 1) >> 1;
            if (bt_row(desc, mid)[3] <= j) lo = mid; else hi = mid - 1;
        }
        const int64_t* D = bt_row(desc, lo);
        const int64_t local = j - D[3];
        const int64_t idx = (int64_t)((const int32_t*)(wire_r + D[6]))[local];
        float* d = dense + D[1] + idx;
        if (accumulate) *d += vals_eval[j];
        else *d = vals_eval[j];
    }
}

std::vector<torch::Tensor> batched_compress_value(
        torch::Tensor values_flat, torch::Tensor desc, torch::Tensor b2t,
        torch::Tensor seg_t, torch::Tensor seg_i, int64_t wire_bytes,
        int64_t k_total, int64_t kmax, int64_t degree, int64_t total_values) {
    CHECK_CUDA(values_flat);
    auto v = values_flat.contiguous();
    auto d = desc.contiguous();
    auto map = b2t.contiguous();
    auto st_t = seg_t.contiguous();
    auto st_i = seg_i.contiguous();
    const int T = (int)d.size(0);
    const int64_t BV = map.numel();
    const int SB = (int)st_t.numel();
    const int d1 = (int)degree + 1;
    auto dev = v.device();
    hipStream_t stream = at::hip::getCurrentHIPStream();

    auto ws = torch::empty({(int64_t)T * (2 * TK_BINS + 4) + 4 * BV},
                           torch::dtype(torch::kInt32).device(dev));
    int* hist1 = ws.data_ptr<int>();
    int* hist2 = hist1 + (int64_t)T * TK_BINS;
    int* sc = hist2 + (int64_t)T * TK_BINS;
    int* counts = sc + (int64_t)T * 4;
    int* offs = counts + 2 * BV;
    auto wire = torch::empty({wire_bytes}, torch::dtype(torch::kUInt8).device(dev));
    auto out_idx = torch::empty({k_total}, torch::dtype(torch::kInt64).device(dev));
    auto vals_tmp = torch::empty({k_total}, torch::dtype(torch::kFloat32).device(dev));
    auto padmat = torch::empty({T, kmax}, torch::dtype(torch::kFloat32).device(dev));
    auto starts = torch::empty({(int64_t)T * PF_SMAX},
                               torch::dtype(torch::kInt64).device(dev));
    zero_ints(hist1, (int64_t)T * (2 * TK_BINS + 4), stream);
    zero_ints((int*)wire.data_ptr<uint8_t>(), wire_bytes / 4, stream);

    const int64_t* dp = d.data_ptr<int64_t>();
    const int* mp = map.data_ptr<int>();
    const float* vp = v.data_ptr<float>();
    hipLaunchKernelGGL(bt_hist_kernel, dim3((int)BV), dim3(TK_BLOCK), 0, stream,
                       vp, dp, mp, 1, hist1, sc);
    hipLaunchKernelGGL(bt_thresh_kernel, dim3(T), dim3(WAVE), 0, stream, hist1, dp, 1, sc);
    hipLaunchKernelGGL(bt_hist_kernel, dim3((int)BV), dim3(TK_BLOCK), 0, stream,
                       vp, dp, mp, 2, hist2, sc);
    hipLaunchKernelGGL(bt_thresh_kernel, dim3(T), dim3(WAVE), 0, stream, hist2, dp, 2, sc);
    hipLaunchKernelGGL(bt_count_kernel, dim3((int)BV), dim3(TK_BLOCK), 0, stream,
                       vp, dp, mp, sc, BV, counts);
    hipLaunchKernelGGL(bt_scan_kernel, dim3(2 * T), dim3(QBLOCK), 0, stream,
                       counts, dp, T, BV, offs);
    hipLaunchKernelGGL(bt_scatter_kernel, dim3((int)BV), dim3(TK_BLOCK), 0, stream,
                       vp, dp, mp, sc, offs, BV, out_idx.data_ptr<int64_t>(),
                       vals_tmp.data_ptr<float>());
    hipLaunchKernelGGL(bt2_fill_ninf, dim3(bt_grid((int64_t)T * kmax)), dim3(256), 0,
                       stream, padmat.data_ptr<float>(), (int64_t)T * kmax);
    hipLaunchKernelGGL(bt2_padmat_kernel, dim3(bt_grid(k_total)), dim3(256), 0, stream,
                       vals_tmp.data_ptr<float>(), dp, T, k_total, kmax,
                       padmat.data_ptr<float>());
    auto sorted_arg = at::sort(padmat, /*stable=*/true, /*dim=*/1, /*descending=*/true);
    auto sorted = std::get<0>(sorted_arg).contiguous();
    auto argsortm = std::get<1>(sorted_arg).contiguous();
    auto num_pos = (sorted > 0).sum(1).to(torch::kFloat64).contiguous();
    hipLaunchKernelGGL(bt2_starts_kernel, dim3(T), dim3(WAVE), 0, stream,
                       num_pos.data_ptr<double>(), dp,
                       starts.data_ptr<int64_t>(), wire.data_ptr<uint8_t>(), d1);
    hipLaunchKernelGGL(bt2_fit_kernel, dim3(SB), dim3(QBLOCK), 0, stream,
                       sorted.data_ptr<float>(), dp, starts.data_ptr<int64_t>(),
                       st_t.data_ptr<int>(), st_i.data_ptr<int>(), (int)degree,
                       kmax, wire.data_ptr<uint8_t>());
    hipLaunchKernelGGL(bt2_write_idx_kernel, dim3(bt_grid(k_total)), dim3(256), 0,
                       stream, out_idx.data_ptr<int64_t>(),
                       argsortm.data_ptr<int64_t>(), dp, T, k_total, kmax,
                       wire.data_ptr<uint8_t>());
    // own decode
    auto vals_eval = torch::empty({k_total}, torch::dtype(torch::kFloat32).device(dev));
    auto own = torch::empty({total_values}, torch::dtype(torch::kFloat32).device(dev));
    hipLaunchKernelGGL(bt_fill_zero_f, dim3(bt_grid(total_values)), dim3(256), 0, stream,
                       own.data_ptr<float>(), total_values);
    hipLaunchKernelGGL(bt2_eval_kernel, dim3(bt_grid(k_total)), dim3(256), 0, stream,
                       wire.data_ptr<uint8_t>(), dp, T, k_total,
                       starts.data_ptr<int64_t>(), d1, vals_eval.data_ptr<float>());
    hipLaunchKernelGGL(bt2_scatter_value_kernel, dim3(bt_grid(k_total)), dim3(256), 0,
                       stream, vals_eval.data_ptr<float>(), wire.data_ptr<uint8_t>(),
                       dp, T, k_total, 0, own.data_ptr<float>());
    return {wire, own};
}

torch::Tensor batched_decode_value_sum(torch::Tensor wires2d, torch::Tensor desc,
                                       int64_t total_values, int64_t k_total,
                                       int64_t degree) {
    CHECK_CUDA(wires2d);
    TORCH_CHECK(wires2d.dim() == 2, "expected [R, W]");
    auto w = wires2d.contiguous();
    auto d = desc.contiguous();
    const int R = (int)w.size(0);
    const int64_t W = w.size(1);
    const int T = (int)d.size(0);
    const int d1 = (int)degree + 1;
    auto dev = w.device();
    hipStream_t stream = at::hip::getCurrentHIPStream();
    auto dense = torch::empty({total_values}, torch::dtype(torch::kFloat32).device(dev));
    auto vals_eval = torch::empty({k_total}, torch::dtype(torch::kFloat32).device(dev));
    auto starts = torch::empty({(int64_t)T * PF_SMAX},
                               torch::dtype(torch::kInt64).device(dev));
    hipLaunchKernelGGL(bt_fill_zero_f, dim3(bt_grid(total_values)), dim3(256), 0, stream,
                       dense.data_ptr<float>(), total_values);
    for (int r = 0; r < R; ++r) {
        const uint8_t* wr = w.data_ptr<uint8_t>() + (int64_t)r * W;
        hipLaunchKernelGGL(bt2_starts_from_wire_kernel, dim3(T), dim3(WAVE), 0, stream,
                           wr, d.data_ptr<int64_t>(), d1, starts.data_ptr<int64_t>());
        hipLaunchKernelGGL(bt2_eval_kernel, dim3(bt_grid(k_total)), dim3(256), 0, stream,
                           wr, d.data_ptr<int64_t>(), T, k_total,
                           starts.data_ptr<int64_t>(), d1, vals_eval.data_ptr<float>());
        hipLaunchKernelGGL(bt2_scatter_value_kernel, dim3(bt_grid(k_total)), dim3(256), 0,
                           stream, vals_eval.data_ptr<float>(), wr,
                           d.data_ptr<int64_t>(), T, k_total, 1,
                           dense.data_ptr<float>());
    }
    return dense;
}

// ---------------------------------------------------------------------------
// CPU-native C++ paths (replace the reference's TF C++ CPU ops:
// bloom_filter_compression.cc / integer_compression.cc) — same wire format
// and hash math as the HIP kernels, parallelized with at::parallel_for.
// ---------------------------------------------------------------------------

#include <ATen/Parallel.h>

// ---------------------------------------------------------------------------
// DoubleExp (Fit-DExp) fused fit kernel — VERDICT r1 item 6.
//
// Reference behavior: tensorflow/deepreduce.py:67-144 (Jacquelin cumulative-
// integral linearization of y = a*e^{bx} + c*e^{dx}): regress y on
// [SS, S, x, 1] where S = cumint(y), SS = cumint(S); b,d are the roots of
// t^2 - B t - A; then a,c from the 2x2 LS of y on (e^{bx}, e^{dx}).
// The torch path chained ~20 fp64 ops (two cumsums, stacks, matmuls, two
// linalg.solve -> 2 hipSolver round trips); here ONE block per tensor does
// both passes: chunked block-scan for S/SS fused with the Gram/moment
// accumulation, in-register Cholesky for the 4x4 and 2x2 solves.
// y must be sorted ascending (the codec sorts by |value|), x = i+1.
// ---------------------------------------------------------------------------

#define DEXP_BLOCK 256

__device__ __forceinline__ double dexp_block_scan(double v, int wid, int lane,
                                                  double* wtot, double* carry) {
    // inclusive block scan of v (one value per thread); carry holds the
    // running prefix from previous chunks and is updated by thread 0
    double incl = v;
    for (int off = 1; off < WAVE; off <<= 1) {
        double up = __shfl_up(incl, off, WAVE);
        if (lane >= off) incl += up;
    }
    if (lane == WAVE - 1) wtot[wid] = incl;
    __syncthreads();
    double wbase = *carry;
    for (int w = 0; w < wid; ++w) wbase += wtot[w];
    double out = wbase + incl;
    __syncthreads();
    if (threadIdx.x == blockDim.x - 1) *carry = out;
    __syncthreads();
    return out;
}

__device__ __forceinline__ void cholesky_solve_inreg(double* G, double* b, int n) {
    // G [n,n] row-major SPD (ridged), solve G x = b in place -> b
    for (int i = 0; i < n; ++i) {
        for (int j_ = 0; j_ <= i; ++j_) {
            double s = G[i * n + j_];
            for (int p = 0; p < j_; ++p) s -= G[i * n + p] * G[j_ * n + p];
            if (i == j_) G[i * n + i] = sqrt(fmax(s, 1e-300));
            else G[i * n + j_] = s / G[j_ * n + j_];
        }
    }
    for (int i = 0; i < n; ++i) {          // forward
        double s = b[i];
        for (int p = 0; p < i; ++p) s -= G[i * n + p] * b[p];
        b[i] = s / G[i * n + i];
    }
    for (int i = n - 1; i >= 0; --i) {     // backward
        double s = b[i];
        for (int p = i + 1; p < n; ++p) s -= G[p * n + i] * b[p];
        b[i] = s / G[i * n + i];
    }
}

__global__ void dexp_fit_kernel(const float* __restrict__ y_flat,
                                const int64_t* __restrict__ offs,
                                const int64_t* __restrict__ lens,
                                double* __restrict__ coeffs /*[B,4]*/) {
    const int b_idx = blockIdx.x;
    const float* __restrict__ y = y_flat + offs[b_idx];
    const int64_t N = lens[b_idx];
    const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;

    __shared__ double wtot[DEXP_BLOCK / WAVE];
    __shared__ double carrS, carrSS, prevy, prevS;
    __shared__ double red[DEXP_BLOCK / WAVE][14];
    __shared__ double theta_s[4];

    if (threadIdx.x == 0) { carrS = 0.0; carrSS = 0.0; prevy = 0.0; prevS = 0.0; }
    __syncthreads();

    double acc[14];
    for (int q = 0; q < 14; ++q) acc[q] = 0.0;

    __shared__ double ylast[DEXP_BLOCK / WAVE];
    __shared__ double Slast[DEXP_BLOCK / WAVE];
    for (int64_t i0 = 0; i0 < N; i0 += blockDim.x) {
        const int64_t i = i0 + threadIdx.x;
        const bool ok = i < N;
        const double yi = ok ? (double)y[i] : 0.0;
        // neighbor y[i-1]: shfl within the wave, LDS across wave/chunk edges
        if (lane == WAVE - 1) ylast[wid] = yi;
        __syncthreads();
        const double upy = __shfl_up(yi, 1, WAVE);
        const double ym1 = (lane > 0) ? upy
                           : ((wid == 0) ? prevy : ylast[wid - 1]);
        const double ds = (i == 0 || !ok) ? 0.0 : 0.5 * (yi + ym1);
        __syncthreads();  // prevy read by all before the update below
        if (ok && threadIdx.x == blockDim.x - 1) prevy = yi;
        const double Si = dexp_block_scan(ds, wid, lane, wtot, &carrS);

        // second trapezoid: (S[i]+S[i-1])/2
        if (lane == WAVE - 1) Slast[wid] = Si;
        __syncthreads();
        const double upS = __shfl_up(Si, 1, WAVE);
        const double Sm1 = (lane > 0) ? upS
                           : ((wid == 0) ? prevS : Slast[wid - 1]);
        const double dss = (i == 0 || !ok) ? 0.0 : 0.5 * (Si + Sm1);
        __syncthreads();
        if (ok && threadIdx.x == blockDim.x - 1) prevS = Si;
        const double SSi = dexp_block_scan(dss, wid, lane, wtot, &carrSS);

        if (ok) {
            const double xi = (double)(i + 1);
            // Gram of [SS, S, x, 1] (10 upper entries) + rhs (4)
            acc[0] += SSi * SSi; acc[1] += SSi * Si; acc[2] += SSi * xi;
            acc[3] += SSi;       acc[4] += Si * Si;  acc[5] += Si * xi;
            acc[6] += Si;        acc[7] += xi * xi;  acc[8] += xi;
            acc[9] += 1.0;
            acc[10] += SSi * yi; acc[11] += Si * yi; acc[12] += xi * yi;
            acc[13] += yi;
        }
        __syncthreads();
    }

    // block-reduce the 14 accumulators
    for (int q = 0; q < 14; ++q) {
        double v = acc[q];
        for (int off = WAVE / 2; off > 0; off >>= 1)
            v += __shfl_down(v, off, WAVE);
        if (lane == 0) red[wid][q] = v;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
        double G[16], rhs[4];
        double s[14];
        for (int q = 0; q < 14; ++q) {
            double v = 0.0;
            for (int w = 0; w < DEXP_BLOCK / WAVE; ++w) v += red[w][q];
            s[q] = v;
        }
        G[0] = s[0];  G[1] = s[1];  G[2] = s[2];  G[3] = s[3];
        G[4] = s[1];  G[5] = s[4];  G[6] = s[5];  G[7] = s[6];
        G[8] = s[2];  G[9] = s[5];  G[10] = s[7]; G[11] = s[8];
        G[12] = s[3]; G[13] = s[6]; G[14] = s[8]; G[15] = s[9];
        rhs[0] = s[10]; rhs[1] = s[11]; rhs[2] = s[12]; rhs[3] = s[13];
        double mx = 0.0;
        for (int q = 0; q < 4; ++q) mx = fmax(mx, fabs(G[q * 4 + q]));
        const double ridge = mx * 1e-12 + 1e-30;
        for (int q = 0; q < 4; ++q) G[q * 4 + q] += ridge;
        cholesky_solve_inreg(G, rhs, 4);
        const double A = rhs[0], B = rhs[1];
        const double disc = fmax(B * B + 4.0 * A, 0.0);
        const double r = sqrt(disc);
        const double cap = 650.0 / fmax((double)N, 1.0);
        theta_s[0] = fmin(fmax(0.5 * (B + r), -cap), cap);   // b
        theta_s[1] = fmin(fmax(0.5 * (B - r), -cap), cap);   // d
    }
    __syncthreads();
    const double bb = theta_s[0], dd = theta_s[1];

    // pass 2: 2x2 LS of y on (e^{b x}, e^{d x})
    double g2[5] = {0, 0, 0, 0, 0};  // ebeb, ebed, eded, eby, edy
    for (int64_t i = threadIdx.x; i < N; i += blockDim.x) {
        const double xi = (double)(i + 1);
        const double yi = (double)y[i];
        const double eb = exp(bb * xi), ed = exp(dd * xi);
        g2[0] += eb * eb; g2[1] += eb * ed; g2[2] += ed * ed;
        g2[3] += eb * yi; g2[4] += ed * yi;
    }
    for (int q = 0; q < 5; ++q) {
        double v = g2[q];
        for (int off = WAVE / 2; off > 0; off >>= 1)
            v += __shfl_down(v, off, WAVE);
        if (lane == 0) red[wid][q] = v;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
        double s2[5];
        for (int q = 0; q < 5; ++q) {
            double v = 0.0;
            for (int w = 0; w < DEXP_BLOCK / WAVE; ++w) v += red[w][q];
            s2[q] = v;
        }
        double G2[4] = {s2[0] + 1e-12, s2[1], s2[1], s2[2] + 1e-12};
        double ac[2] = {s2[3], s2[4]};
        cholesky_solve_inreg(G2, ac, 2);
        coeffs[(int64_t)b_idx * 4 + 0] = ac[0];
        coeffs[(int64_t)b_idx * 4 + 1] = bb;
        coeffs[(int64_t)b_idx * 4 + 2] = ac[1];
        coeffs[(int64_t)b_idx * 4 + 3] = dd;
    }
}

torch::Tensor dexp_fit(torch::Tensor y, torch::Tensor offs, torch::Tensor lens) {
    CHECK_CUDA(y);
    auto yy = y.contiguous();
    auto o = offs.contiguous();
    auto l = lens.contiguous();
    TORCH_CHECK(yy.dtype() == torch::kFloat32, "y must be float32");
    const int B = (int)o.numel();
    auto out = torch::empty({B, 4}, torch::dtype(torch::kFloat64).device(y.device()));
    hipStream_t stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(dexp_fit_kernel, dim3(B), dim3(DEXP_BLOCK), 0, stream,
                       yy.data_ptr<float>(), o.data_ptr<int64_t>(),
                       l.data_ptr<int64_t>(), out.data_ptr<double>());
    return out;
}

// ---------------------------------------------------------------------------
// Block-PFoR pack/unpack kernels (codecs/intpack.py wire format v2:
// 128-int blocks, per-block width, patched exceptions, 4-byte-aligned
// regions).  The torch implementation stays as the CPU path; these kernels
// produce BYTE-IDENTICAL wires (roundtrip + device-independence tested).
// One wave per block for stats/exceptions; one thread per output word /
// per value for the streams.
// ---------------------------------------------------------------------------

#define PFOR_BLK 128

__device__ __forceinline__ int pfor_bitlen(int64_t v) {
    return v > 0 ? 64 - __clzll((uint64_t)v) : 0;
}

// one wave per 128-int block: choose width, count exceptions, emit sizes
__global__ void pfor_stats_kernel(const int64_t* __restrict__ v, int64_t n,
                                  int64_t nb, int* __restrict__ width,
                                  int* __restrict__ nexc,
                                  int* __restrict__ excw,
                                  int* __restrict__ sbytes /*aligned*/,
                                  int* __restrict__ ebytes /*aligned*/) {
    const int64_t blk = blockIdx.x;
    if (blk >= nb) return;
    const int lane = threadIdx.x;  // 64 lanes
    const int64_t base = blk * PFOR_BLK;
    const int cnt = (int)min((int64_t)PFOR_BLK, n - base);
    // lane -> (j, j+64): within each half, ballot lane order == ascending
    // j, so exception ranks match the torch encoder's flat enumeration
    const int j0 = lane, j1 = lane + WAVE;
    const int bl0 = (j0 < cnt) ? pfor_bitlen(v[base + j0]) : 0;
    const int bl1 = (j1 < cnt) ? pfor_bitlen(v[base + j1]) : 0;
    // wave max bitlen
    int mx = max(bl0, bl1);
    for (int off = WAVE / 2; off > 0; off >>= 1)
        mx = max(mx, __shfl_down(mx, off, WAVE));
    mx = __shfl(mx, 0, WAVE);
    // cost(b) = ceil(cnt*b/8) + ne + ceil(ne*exw/8) (same model as the
    // torch encoder, UNALIGNED, so argmin matches bit-for-bit)
    int best_b = 0, best_cost = 0x7FFFFFFF, best_ne = 0;
    for (int b = 0; b <= 32; ++b) {
        const int ne = __popcll(__ballot(bl0 > b)) + __popcll(__ballot(bl1 > b));
        const int exw = mx > b ? mx - b : 0;
        const int cost = (cnt * b + 7) / 8 + ne + (ne * exw + 7) / 8;
        if (cost < best_cost) { best_cost = cost; best_b = b; best_ne = ne; }
    }
    if (lane == 0) {
        const int exw = (best_ne > 0 && mx > best_b) ? mx - best_b : 0;
        width[blk] = best_b;
        nexc[blk] = best_ne;
        excw[blk] = exw;
        sbytes[blk] = (((cnt * best_b + 7) / 8) + 3) & ~3;
        ebytes[blk] = ((best_ne + 3) & ~3) + ((((best_ne * exw + 7) / 8) + 3) & ~3);
    }
}

// one thread per 32-bit output word of the low-bit stream
__global__ void pfor_pack_kernel(const int64_t* __restrict__ v, int64_t n,
                                 int64_t nb, const int* __restrict__ width,
                                 const int64_t* __restrict__ sw_off /*words*/,
                                 int64_t total_sw, uint32_t* __restrict__ stream) {
    int64_t w = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; w < total_sw; w += stride) {
        int lo = 0, hi = (int)nb - 1;      // largest blk with sw_off[blk] <= w
        while (lo < hi) {
            const int mid = (lo + hi + 1) >> 1;
            if (sw_off[mid] <= w) lo = mid; else hi = mid - 1;
        }
        const int b = width[lo];
        uint32_t out = 0;
        if (b > 0) {
            const int64_t base = (int64_t)lo * PFOR_BLK;
            const int cnt = (int)min((int64_t)PFOR_BLK, n - base);
            const int64_t wbit = (w - sw_off[lo]) * 32;   // bit off in block
            const uint64_t mask = (b >= 64) ? ~0ull : ((1ull << b) - 1);
            int j = (int)(wbit / b);
            for (; j < cnt && (int64_t)j * b < wbit + 32; ++j) {
                const uint64_t val = (uint64_t)v[base + j] & mask;
                const int64_t sh = (int64_t)j * b - wbit;  // may be negative
                if (sh >= 0) out |= (uint32_t)(val << sh);
                else out |= (uint32_t)(val >> (-sh));
            }
        }
        stream[w] = out;
    }
}

// one wave per block: exception positions (bytes) + high bits (atomicOr)
__global__ void pfor_exc_kernel(const int64_t* __restrict__ v, int64_t n,
                                int64_t nb, const int* __restrict__ width,
                                const int* __restrict__ nexc,
                                const int* __restrict__ excw,
                                const int64_t* __restrict__ e_off /*bytes*/,
                                uint8_t* __restrict__ exc) {
    const int64_t blk = blockIdx.x;
    if (blk >= nb) return;
    const int lane = threadIdx.x;
    const int b = width[blk], ne = nexc[blk], exw = excw[blk];
    if (ne == 0) return;
    const int64_t base = blk * PFOR_BLK;
    const int cnt = (int)min((int64_t)PFOR_BLK, n - base);
    uint8_t* pos_out = exc + e_off[blk];
    uint32_t* hi_out = (uint32_t*)(exc + e_off[blk] + ((ne + 3) & ~3));
    int rank_base = 0;
    for (int half = 0; half < 2; ++half) {
        const int j = lane + half * WAVE;
        const bool is_exc = (j < cnt) && (pfor_bitlen(v[base + j]) > b);
        const uint64_t ball = __ballot(is_exc);
        if (is_exc) {
            const uint64_t below = (lane == 63) ? (~0ull >> 1)
                                                : ((1ull << lane) - 1);
            const int rank = rank_base + __popcll(ball & below);
            pos_out[rank] = (uint8_t)j;
            if (exw > 0) {
                const uint64_t hiv = (uint64_t)v[base + j] >> b;
                const int64_t sbit = (int64_t)rank * exw;
                const int wi = (int)(sbit >> 5), sh = (int)(sbit & 31);
                atomicOr(&hi_out[wi], (uint32_t)(hiv << sh));
                if (sh + exw > 32)
                    atomicOr(&hi_out[wi + 1], (uint32_t)(hiv >> (32 - sh)));
            }
        }
        rank_base += __popcll(ball);
    }
}

// one thread per value: gather its b low bits from <=2 stream words
__global__ void pfor_unpack_kernel(const uint32_t* __restrict__ stream,
                                   int64_t n, const int* __restrict__ width,
                                   const int64_t* __restrict__ sw_off,
                                   int64_t* __restrict__ out) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        const int64_t blk = i / PFOR_BLK;
        const int b = width[blk];
        if (b == 0) { out[i] = 0; continue; }
        const int j = (int)(i - blk * PFOR_BLK);
        const int64_t sbit = (int64_t)j * b;
        const int64_t wi = sw_off[blk] + (sbit >> 5);
        const int sh = (int)(sbit & 31);
        uint64_t bits = (uint64_t)stream[wi] >> sh;
        if (sh + b > 32) bits |= (uint64_t)stream[wi + 1] << (32 - sh);
        out[i] = (int64_t)(bits & ((b >= 64) ? ~0ull : ((1ull << b) - 1)));
    }
}

// one wave per block: apply exception high bits
__global__ void pfor_unexc_kernel(const uint8_t* __restrict__ exc, int64_t n,
                                  int64_t nb, const int* __restrict__ width,
                                  const int* __restrict__ nexc,
                                  const int* __restrict__ excw,
                                  const int64_t* __restrict__ e_off,
                                  int64_t* __restrict__ out) {
    const int64_t blk = blockIdx.x;
    if (blk >= nb) return;
    const int b = width[blk], ne = nexc[blk], exw = excw[blk];
    if (ne == 0 || exw == 0) return;
    const uint8_t* pos_in = exc + e_off[blk];
    const uint32_t* hi_in = (const uint32_t*)(exc + e_off[blk] + ((ne + 3) & ~3));
    for (int r = threadIdx.x; r < ne; r += blockDim.x) {
        const int j = pos_in[r];
        const int64_t sbit = (int64_t)r * exw;
        const int wi = (int)(sbit >> 5), sh = (int)(sbit & 31);
        uint64_t bits = (uint64_t)hi_in[wi] >> sh;
        if (sh + exw > 32) bits |= (uint64_t)hi_in[wi + 1] << (32 - sh);
        bits &= (exw >= 64) ? ~0ull : ((1ull << exw) - 1);
        out[blk * PFOR_BLK + j] |= (int64_t)(bits << b);
    }
}

torch::Tensor pfor_pack(torch::Tensor ints) {
    CHECK_CUDA(ints);
    auto v = ints.to(torch::kInt64).contiguous();
    const int64_t n = v.numel();
    const int64_t nb = (n + PFOR_BLK - 1) / PFOR_BLK;
    auto dev = v.device();
    hipStream_t stream = at::hip::getCurrentHIPStream();
    auto opts32 = torch::dtype(torch::kInt32).device(dev);
    auto header = torch::tensor(
        {(int)(n & 255), (int)((n >> 8) & 255), (int)((n >> 16) & 255),
         (int)((n >> 24) & 255), 0xFE, (int)(nb & 255), (int)((nb >> 8) & 255),
         (int)((nb >> 16) & 255), (int)((nb >> 24) & 255)},
        torch::dtype(torch::kUInt8)).to(dev);
    if (n == 0) return header;

    auto stats = torch::empty({5 * nb}, opts32);
    int* width = stats.data_ptr<int>();
    int* nexc = width + nb;
    int* excw = nexc + nb;
    int* sbytes = excw + nb;
    int* ebytes = sbytes + nb;
    hipLaunchKernelGGL(pfor_stats_kernel, dim3((int)nb), dim3(WAVE), 0, stream,
                       v.data_ptr<int64_t>(), n, nb, width, nexc, excw,
                       sbytes, ebytes);
    auto sb64 = stats.narrow(0, 3 * nb, nb).to(torch::kInt64);
    auto eb64 = stats.narrow(0, 4 * nb, nb).to(torch::kInt64);
    auto sw_off = (sb64.cumsum(0) - sb64).floor_divide(4);  // word offsets
    auto e_off = eb64.cumsum(0) - eb64;
    const int64_t total_sb = sb64.sum().item<int64_t>();   // host sync
    const int64_t total_eb = eb64.sum().item<int64_t>();
    auto sw_off_c = sw_off.contiguous();
    auto e_off_c = e_off.contiguous();

    // meta bytes [b, ne, exw] per block + pad to 4
    auto meta = torch::stack({stats.narrow(0, 0, nb), stats.narrow(0, nb, nb),
                              stats.narrow(0, 2 * nb, nb)}, 1)
                    .to(torch::kUInt8).reshape({-1});
    auto headmeta = torch::cat({header, meta});
    const int64_t hm_pad = (4 - (headmeta.numel() & 3)) & 3;
    if (hm_pad)
        headmeta = torch::cat({headmeta,
                               torch::zeros({hm_pad}, torch::dtype(torch::kUInt8).device(dev))});

    auto body = torch::zeros({total_sb + total_eb},
                             torch::dtype(torch::kUInt8).device(dev));
    uint32_t* sw = (uint32_t*)body.data_ptr<uint8_t>();
    uint8_t* exc = body.data_ptr<uint8_t>() + total_sb;
    hipLaunchKernelGGL(pfor_pack_kernel, dim3(bt_grid(total_sb / 4)), dim3(256),
                       0, stream, v.data_ptr<int64_t>(), n, nb, width,
                       sw_off_c.data_ptr<int64_t>(), total_sb / 4, sw);
    hipLaunchKernelGGL(pfor_exc_kernel, dim3((int)nb), dim3(WAVE), 0, stream,
                       v.data_ptr<int64_t>(), n, nb, width, nexc, excw,
                       e_off_c.data_ptr<int64_t>(), exc);
    return torch::cat({headmeta, body});
}

torch::Tensor pfor_unpack(torch::Tensor wire) {
    CHECK_CUDA(wire);
    auto w = wire.contiguous();
    auto head = w.narrow(0, 0, 9).cpu();
    const uint8_t* h = head.data_ptr<uint8_t>();
    const int64_t n = (int64_t)h[0] | ((int64_t)h[1] << 8) |
                      ((int64_t)h[2] << 16) | ((int64_t)h[3] << 24);
    TORCH_CHECK(h[4] == 0xFE, "not a block-PFoR v2 wire");
    const int64_t nb = (int64_t)h[5] | ((int64_t)h[6] << 8) |
                       ((int64_t)h[7] << 16) | ((int64_t)h[8] << 24);
    auto dev = w.device();
    auto out = torch::zeros({n}, torch::dtype(torch::kInt64).device(dev));
    if (n == 0) return out;
    hipStream_t stream = at::hip::getCurrentHIPStream();

    auto meta = w.narrow(0, 9, nb * 3).to(torch::kInt32).reshape({nb, 3});
    auto width_t = meta.select(1, 0).contiguous();
    auto ne_t = meta.select(1, 1).contiguous();
    auto exw_t = meta.select(1, 2).contiguous();
    auto cnt = torch::full({nb}, (int64_t)PFOR_BLK,
                           torch::dtype(torch::kInt64).device(dev));
    if (n % PFOR_BLK) cnt[nb - 1].fill_(n % PFOR_BLK);
    auto w64 = width_t.to(torch::kInt64);
    auto ne64 = ne_t.to(torch::kInt64);
    auto exw64 = exw_t.to(torch::kInt64);
    auto sb = ((cnt * w64 + 7).floor_divide(8) + 3).bitwise_and(~3);
    auto eb = (ne64 + 3).bitwise_and(~3) +
              ((ne64 * exw64 + 7).floor_divide(8) + 3).bitwise_and(~3);
    auto sw_off = (sb.cumsum(0) - sb).floor_divide(4).contiguous();
    auto e_off = (eb.cumsum(0) - eb).contiguous();
    const int64_t total_sb = sb.sum().item<int64_t>();
    int64_t hm = 9 + nb * 3;
    hm += (4 - (hm & 3)) & 3;

    auto body = w.narrow(0, hm, w.numel() - hm).contiguous();
    const uint32_t* sw = (const uint32_t*)body.data_ptr<uint8_t>();
    const uint8_t* exc = body.data_ptr<uint8_t>() + total_sb;
    hipLaunchKernelGGL(pfor_unpack_kernel, dim3(bt_grid(n)), dim3(256), 0,
                       stream, sw, n, width_t.data_ptr<int>(),
                       sw_off.data_ptr<int64_t>(), out.data_ptr<int64_t>());
    hipLaunchKernelGGL(pfor_unexc_kernel, dim3((int)nb), dim3(WAVE), 0, stream,
                       exc, n, nb, width_t.data_ptr<int>(),
                       ne_t.data_ptr<int>(), exw_t.data_ptr<int>(),
                       e_off.data_ptr<int64_t>(), out.data_ptr<int64_t>());
    return out;
}

torch::Tensor bloom_insert_cpu(torch::Tensor idxs, int64_t m, int64_t num_hash) {
    auto items = idxs.to(torch::kInt64).contiguous();
    int64_t nbytes = ceil_div(m, 8);
    auto out = torch::zeros({nbytes}, torch::dtype(torch::kUInt8));
    uint8_t* bits = out.data_ptr<uint8_t>();
    const int64_t* it = items.data_ptr<int64_t>();
    int64_t n = items.numel();
    // serial (insert is tiny: k items, k*num_hash bit sets)
    for (int64_t i = 0; i < n; ++i) {
        uint32_t h1, h2;
        hash_bases(it[i], &h1, &h2);
        uint32_t x = h1;
        for (int64_t j = 0; j < num_hash; ++j, x += h2) {
            uint64_t pos = bloom_pos(x, m);
            bits[pos >> 3] |= (uint8_t)(1u << (pos & 7));
        }
    }
    return out;
}

torch::Tensor bloom_query_positives_cpu(torch::Tensor packed, int64_t m, int64_t num_hash,
                                        int64_t universe) {
    auto p = packed.contiguous();
    const uint8_t* bits = p.data_ptr<uint8_t>();
    const int64_t chunk = 1 << 16;
    int64_t nchunks = ceil_div(universe, chunk);
    std::vector<std::vector<int64_t>> found((size_t)nchunks);
    at::parallel_for(0, nchunks, 1, [&](int64_t c0, int64_t c1) {
        for (int64_t c = c0; c < c1; ++c) {
            int64_t start = c * chunk, end = std::min(start + chunk, universe);
            auto& v = found[(size_t)c];
            for (int64_t i = start; i < end; ++i)
                if (bloom_test(bits, m, (int)num_hash, i)) v.push_back(i);
        }
    });
    int64_t total = 0;
    for (auto& v : found) total += (int64_t)v.size();
    auto out = torch::empty({total}, torch::dtype(torch::kInt64));
    int64_t* o = out.data_ptr<int64_t>();
    for (auto& v : found) {
        std::copy(v.begin(), v.end(), o);
        o += v.size();
    }
    return out;
}

torch::Tensor bloom_query_members_cpu(torch::Tensor packed, int64_t m, int64_t num_hash,
                                      torch::Tensor items) {
    auto p = packed.contiguous();
    auto it = items.to(torch::kInt64).contiguous();
    const uint8_t* bits = p.data_ptr<uint8_t>();
    const int64_t* iv = it.data_ptr<int64_t>();
    int64_t n = it.numel();
    auto out = torch::empty({n}, torch::dtype(torch::kBool));
    bool* o = out.data_ptr<bool>();
    at::parallel_for(0, n, 4096, [&](int64_t a, int64_t b) {
        for (int64_t i = a; i < b; ++i) o[i] = bloom_test(bits, m, (int)num_hash, iv[i]);
    });
    return out;
}

torch::Tensor pack_ints_cpu(torch::Tensor values, int64_t nbits) {
    auto v = values.to(torch::kInt64).contiguous();
    int64_t n = v.numel();
    int64_t nbytes = ceil_div(n * nbits, 8);
    auto out = torch::zeros({nbytes}, torch::dtype(torch::kUInt8));
    uint8_t* o = out.data_ptr<uint8_t>();
    const int64_t* vv = v.data_ptr<int64_t>();
    for (int64_t i = 0; i < n; ++i) {
        int64_t bit0 = i * nbits;
        uint64_t val = (uint64_t)vv[i];
        for (int64_t b = 0; b < nbits; ++b) {
            int64_t bit = bit0 + b;
            o[bit >> 3] |= (uint8_t)(((val >> b) & 1) << (bit & 7));
        }
    }
    return out;
}


torch::Tensor huffman_decode_cpu(torch::Tensor stream, int64_t n,
                                 torch::Tensor codes, torch::Tensor lengths) {
    // canonical Huffman decode of n byte-symbols (codecs/huffman.py tables):
    // replaces the per-bit python loop (1.1 ms at n=1.5k) with ~ns/symbol
    auto st = stream.contiguous();
    auto cd = codes.to(torch::kInt64).contiguous();
    auto ln = lengths.to(torch::kInt64).contiguous();
    const uint8_t* bytes = st.data_ptr<uint8_t>();
    const int64_t* C = cd.data_ptr<int64_t>();
    const int64_t* L = ln.data_ptr<int64_t>();
    // per-length tables
    int64_t first_code[64], first_rank[64], max_end[64];
    for (int i = 0; i < 64; ++i) { first_code[i] = -1; max_end[i] = -1; first_rank[i] = 0; }
    // canonical order: (length, symbol)
    std::vector<int> order;
    for (int s2 = 0; s2 < 256; ++s2)
        if (L[s2] > 0) order.push_back(s2);
    std::stable_sort(order.begin(), order.end(),
                     [&](int a, int b) { return L[a] != L[b] ? L[a] < L[b] : a < b; });
    std::vector<uint8_t> sym_by_rank(order.begin(), order.end());
    int64_t code = 0, prev_len = 0, rank = 0;
    for (int s2 : order) {
        int64_t len = L[s2];
        code <<= (len - prev_len);
        if (first_code[len] < 0) { first_code[len] = code; first_rank[len] = rank; }
        prev_len = len;
        max_end[len] = ++code;
        ++rank;
    }
    auto out = torch::empty({n}, torch::dtype(torch::kUInt8));
    uint8_t* o = out.data_ptr<uint8_t>();
    int64_t pos = 0;  // bit position, LSB-first within each byte
    for (int64_t i = 0; i < n; ++i) {
        int64_t acc = 0, len = 0;
        for (;;) {
            acc = (acc << 1) | ((bytes[pos >> 3] >> (pos & 7)) & 1);
            ++pos;
            ++len;
            if (first_code[len] >= 0 && acc < max_end[len]) {
                o[i] = sym_by_rank[(size_t)(first_rank[len] + (acc - first_code[len]))];
                break;
            }
        }
    }
    return out;
}

torch::Tensor unpack_ints_cpu(torch::Tensor stream, int64_t n, int64_t nbits) {
    auto s = stream.contiguous();
    const uint8_t* sv = s.data_ptr<uint8_t>();
    auto out = torch::empty({n}, torch::dtype(torch::kInt64));
    int64_t* o = out.data_ptr<int64_t>();
    at::parallel_for(0, n, 8192, [&](int64_t a, int64_t b) {
        for (int64_t i = a; i < b; ++i) {
            int64_t bit0 = i * nbits;
            uint64_t acc = 0;
            int need = (int)((nbits + (bit0 & 7) + 7) / 8);
            for (int k = 0; k < need; ++k) acc |= (uint64_t)sv[(bit0 >> 3) + k] << (8 * k);
            o[i] = (int64_t)((acc >> (bit0 & 7)) &
                             ((nbits == 64) ? ~0ull : ((1ull << nbits) - 1)));
        }
    });
    return out;
}

// ---------------------------------------------------------------------------

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("bloom_insert", &bloom_insert, "Bloom insert (HIP)");
    m.def("bloom_query_positives", &bloom_query_positives, "Bloom full-universe query (HIP)");
    m.def("bloom_query_positives_multi", &bloom_query_positives_multi,
          "Batched multi-rank Bloom query (HIP): hash once, test R filters");
    m.def("bloom_query_leftmost", &bloom_query_leftmost,
          "Sync-free first-k-positives query, [R, k] (HIP)");
    m.def("bloom_query_members", &bloom_query_members, "Bloom membership test (HIP)");
    m.def("qsgd_quantize", &qsgd_quantize, "QSGD quantize (HIP)");
    m.def("qsgd_dequantize", &qsgd_dequantize, "QSGD dequantize (HIP)");
    m.def("pack_ints", &pack_ints, "n-bit pack (HIP)");
    m.def("unpack_ints", &unpack_ints, "n-bit unpack (HIP)");
    m.def("topk_select", &topk_select, "histogram-threshold top-k (HIP)");
    m.def("cholesky_solve_small", &cholesky_solve_small, "batched tiny SPD solve (HIP)");
    m.def("batched_compress", &batched_compress,
          "whole-model fused topk+bloom+query+gather -> (wire, out_idx)");
    m.def("batched_scatter_dense", &batched_scatter_dense,
          "own-payload decode: (wire, out_idx) -> dense flat");
    m.def("batched_decode_sum", &batched_decode_sum,
          "multi-rank decode: [R, W] wires -> sum of dense decodes");
    m.def("polyfit_fit", &polyfit_fit,
          "fused piecewise polynomial fit: moments+gram+cholesky per segment");
    m.def("polyfit_eval", &polyfit_eval, "fused piecewise Horner eval");
    m.def("polyfit_starts", &polyfit_starts,
          "device-side padded segment boundaries from num_pos");
    m.def("batched_compress_both", &batched_compress_both,
          "whole-model 'both' compress: bloom+polyfit+packed mapping -> (wire, own_dense)");
    m.def("batched_decode_both_sum", &batched_decode_both_sum,
          "multi-rank 'both' decode: [R, W] wires -> sum of dense decodes");
    m.def("batched_compress_value", &batched_compress_value,
          "whole-model value-mode compress: polyfit coeffs + int32 idxs");
    m.def("batched_decode_value_sum", &batched_decode_value_sum,
          "multi-rank value-mode decode");
    m.def("pfor_pack", &pfor_pack, "block-PFoR encode (HIP)");
    m.def("pfor_unpack", &pfor_unpack, "block-PFoR decode (HIP)");
    m.def("dexp_fit", &dexp_fit,
          "fused DoubleExp cumulative-integral fit (one block per tensor)");
    m.def("bloom_insert_cpu", &bloom_insert_cpu, "Bloom insert (C++ CPU)");
    m.def("bloom_query_positives_cpu", &bloom_query_positives_cpu, "Bloom query (C++ CPU)");
    m.def("bloom_query_members_cpu", &bloom_query_members_cpu, "Bloom members (C++ CPU)");
    m.def("pack_ints_cpu", &pack_ints_cpu, "n-bit pack (C++ CPU)");
    m.def("unpack_ints_cpu", &unpack_ints_cpu, "n-bit unpack (C++ CPU)");
    m.def("huffman_decode_cpu", &huffman_decode_cpu, "canonical Huffman decode (C++ CPU)");
}
