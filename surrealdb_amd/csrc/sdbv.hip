// sdbv.hip — MI355X-native (gfx950/CDNA4) implementation of the SurrealDB
// vector-KNN hot path behind the C-ABI declared in include/sdbv.h.
//
// THIS IS THE PRODUCT PATH. It never calls into oracle/ (the parity oracle is
// test infrastructure); results are defined by the same restated reference
// contracts:
//  - cosine distance: Distance::calculate F32 semantics
//    (surrealdb/core/src/idx/trees/vector.rs:244-249): f32 dot with the
//    ndarray 0.17.2 eightfold-unrolled accumulation, f32 sum-of-squares norms,
//    f64 finish 1 - dot/(|a||b|).
//  - euclidean: vector.rs:282-283 via ndarray-stats l2_dist: sequential f32
//    (a-b)^2 accumulation, f64 sqrt.
//  - result order: ascending (f64 total_cmp distance, id) — knn.rs:128-160
//    FloatKey + knn.rs:363 BTreeSet<(FloatKey, VectorId)>; equals KnnTopK's
//    insertion-order tie-break (knn_topk.rs:61-73) for monotone ids.
//
// Design (MI355X-first, see DESIGN.md for the full rationale):
//  - A table's vectors live in HBM FEATURE-MAJOR (column-major [d][n_pad]):
//    one lane owns one corpus row, a wave's 64 lanes own 64 consecutive rows,
//    so every k-step loads 64x4 B = one fully-coalesced 256 B line (float4 =
//    4 rows/lane = 1 KiB per instruction). Per-lane accumulation preserves
//    the reference's per-row summation order BIT-EXACTLY while the chip
//    streams at the HBM roofline — no cross-lane reduction anywhere on the
//    scan path.
//  - The scan is HBM-read bound (3072 B/row vs ~1.5 kFLOP/row at d=768);
//    MFMA would not help a single query. The batched-query path
//    (sdbv_knn_batch) is the genuinely-dense case and uses MFMA tiles.
//  - Top-K: per-block threshold + LDS candidate buffer + single-wave
//    (dist,id)-exact selection; a final single-block kernel merges per-block
//    winners. Selection is by value, so results are deterministic and
//    scheduling-independent.
//
// Compile: hipcc --offload-arch=gfx950 -O3 (NO -ffast-math; accumulation
// chains use __fmul_rn/__fadd_rn so the compiler cannot contract them into
// fma — the reference's mul-then-add rounding is the contract).

#include <hip/hip_runtime.h>
#include <rocblas/rocblas.h>

#include <algorithm>
#include <atomic>
#include <cfloat>
#include <chrono>
#include <cmath>
#include <cstdint>
#include <deque>
#include <limits>
#include <memory>
#include <thread>
#include <unordered_set>
#include <cstdio>
#include <cstring>
#include <map>
#include <mutex>
#include <queue>
#include <set>
#include <string>
#include <unordered_map>
#include <vector>

#include "../../include/sdbv.h"

#define THREADS 256
#define ROWS_PER_LANE 4
#define TILE (THREADS * ROWS_PER_LANE) /* 1024 rows per block-sweep */
#define MAX_K 64
#define MAX_D 4096 /* query staged in LDS: 16 KiB cap */

// ---------------------------------------------------------------------------
// Error plumbing
// ---------------------------------------------------------------------------
#define HIP_CHECK(ctx, call)                                                   \
	do {                                                                       \
		hipError_t _e = (call);                                                \
		if (_e != hipSuccess) {                                                \
			(ctx)->err = std::string(#call) + ": " + hipGetErrorString(_e);    \
			return SDBV_ERR_HIP;                                               \
		}                                                                      \
	} while (0)

struct Table {
	uint64_t n = 0;
	uint64_t n_pad = 0;
	uint32_t d = 0;
	uint8_t metric = 0;
	double order = 0.0;       // minkowski order (sdbv_table_set_order)
	float *cm = nullptr;      // [d][n_pad] feature-major
	double *norms = nullptr;  // per-row f64 norm (cosine only)
	float *aux = nullptr;     // per-row f32 selection key aux for the batch
	                          // path: cosine 1/norm, euclidean sum-of-squares
	uint64_t *ids_dev = nullptr; // per-row id, device (merge kernel maps)
	uint64_t bytes = 0;
};

struct sdbv_ctx {
	int device = 0;
	hipStream_t stream = nullptr;
	std::map<uint64_t, Table> tables;
	std::mutex mu;
	std::string err;
	sdbv_stats stats{};
	// scratch
	void *block_out = nullptr; // [max_blocks][MAX_K] Cand
	uint64_t block_out_cap = 0;
	void *final_out = nullptr; // [MAX_K] Cand + ids
	void *merge_tmp = nullptr; // [32][MAX_K] Cand (tree-merge level 1)
	float *q_dev = nullptr;
	float *q_pin = nullptr; // pinned H2D staging for the per-query upload
	uint32_t q_cap = 0;
	hipEvent_t ev0, ev1, ev2;
	// batched path
	rocblas_handle blas = nullptr;
	float *S = nullptr; // chunk scores scratch, [chunk][b] col-major
	uint64_t S_cap = 0;
	void *bstate = nullptr; // [b][kk] running candidates
	uint64_t bstate_cap = 0;
	float *Q_dev = nullptr;
	uint64_t Q_cap = 0;
	double *qnorms = nullptr;
	uint64_t qnorms_cap = 0;
	double *bdists = nullptr; // [b][kk] exact distances
	uint64_t bdists_cap = 0;
	// fused MFMA batch path scratch
	uint32_t *btheta = nullptr; // [b] per-query kth-best u32 key
	uint64_t btheta_cap = 0;
	void *bcand = nullptr; // [b][FMM_CAND_CAP] BCand survivors
	uint64_t bcand_cap = 0;
	uint32_t *bcand_cnt = nullptr; // [b]
	uint64_t bcand_cnt_cap = 0;
	double ms_gemm = 0, ms_select = 0, ms_exact = 0; // last batch timings
};

struct Cand {
	double dist;
	uint64_t id; // staged id (already mapped via ids_dev)
};

// ---------------------------------------------------------------------------
// Device helpers
// ---------------------------------------------------------------------------
__device__ __host__ static inline uint64_t d_splitmix64(uint64_t z) {
	z += 0x9E3779B97F4A7C15ULL;
	z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
	z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
	return z ^ (z >> 31);
}

// The committed synthetic-data contract — bit-identical to oracle orc_gen_elem.
__device__ __host__ static inline float d_gen_elem(uint64_t seed, uint64_t gidx) {
	uint64_t x = d_splitmix64(seed + gidx);
	double u = (double)(x >> 11) * 0x1.0p-53;
	return (float)(-20.0 + 40.0 * u);
}

// f64 total_cmp key (knn.rs:128-160): monotone u64.
__device__ static inline uint64_t d_total_key(double x) {
	uint64_t bits = __double_as_longlong(x);
	return (bits >> 63) ? ~bits : (bits | 0x8000000000000000ULL);
}

// ---------------------------------------------------------------------------
// Staging kernels
// ---------------------------------------------------------------------------

// Fill feature-major store with synthetic data: cm[k][r] = elem(row_offset+r, k).
__global__ void k_gen_cm(float *cm, uint64_t n, uint64_t n_pad, uint32_t d,
                         uint64_t seed, uint64_t row_offset) {
	uint64_t total = (uint64_t)d * n_pad;
	for (uint64_t idx = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
	     idx < total; idx += (uint64_t)gridDim.x * blockDim.x) {
		uint64_t k = idx / n_pad;
		uint64_t r = idx % n_pad;
		float v = 0.0f;
		if (r < n)
			v = d_gen_elem(seed, (row_offset + r) * (uint64_t)d + k);
		cm[idx] = v;
	}
}

// Transpose row-major [n][d] -> feature-major [d][n_pad], LDS-tiled 32x32.
__global__ void k_transpose(const float *__restrict__ rm, float *__restrict__ cm,
                            uint64_t n, uint64_t n_pad, uint32_t d) {
	__shared__ float tile[32][33];
	uint64_t rb = (uint64_t)blockIdx.x * 32; // row base
	uint32_t kb = blockIdx.y * 32;           // feature base
	uint32_t tx = threadIdx.x & 31, ty = threadIdx.x >> 5; // 32x8
	for (uint32_t yy = ty; yy < 32; yy += 8) {
		uint64_t r = rb + yy;
		uint32_t k = kb + tx;
		tile[yy][tx] = (r < n && k < d) ? rm[r * d + k] : 0.0f;
	}
	__syncthreads();
	for (uint32_t yy = ty; yy < 32; yy += 8) {
		uint32_t k = kb + yy;
		uint64_t r = rb + tx;
		if (k < d && r < n_pad)
			cm[(uint64_t)k * n_pad + r] = tile[tx][yy];
	}
}

// Per-row norms for cosine (vector.rs:246-247): sqrt(f64(sum a*a)) with the
// restated unrolled_fold f32 chain. One lane per row, feature-major reads.
__global__ void k_norms(const float *__restrict__ cm, double *__restrict__ norms,
                        uint64_t n, uint64_t n_pad, uint32_t d) {
	uint64_t r = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
	if (r >= n)
		return;
	float p[8] = {0, 0, 0, 0, 0, 0, 0, 0};
	uint32_t k = 0;
	for (; k + 8 <= d; k += 8) {
#pragma unroll
		for (uint32_t j = 0; j < 8; j++) {
			float x = cm[(uint64_t)(k + j) * n_pad + r];
			p[j] = __fadd_rn(p[j], __fmul_rn(x, x));
		}
	}
	float acc = 0.0f;
	acc = __fadd_rn(acc, __fadd_rn(__fadd_rn(p[0], p[4]), __fadd_rn(p[1], p[5])));
	acc = __fadd_rn(acc, __fadd_rn(__fadd_rn(p[2], p[6]), __fadd_rn(p[3], p[7])));
	for (; k < d; k++) {
		float x = cm[(uint64_t)k * n_pad + r];
		acc = __fadd_rn(acc, __fmul_rn(x, x));
	}
	norms[r] = sqrt((double)acc);
}

__global__ void k_fill_ids(uint64_t *ids, uint64_t n, uint64_t id_base) {
	uint64_t r = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
	if (r < n)
		ids[r] = id_base + r;
}

// ---------------------------------------------------------------------------
// Brute-force scan kernel.
// Each block owns a contiguous row span; per 1024-row tile each lane computes
// ROWS_PER_LANE distances with the reference's per-row accumulation chain,
// threshold-checks against the block's current kth-best, and appends rare
// survivors to an LDS candidate buffer; wave 0 then merges candidates into
// the block top-K by exact (total_key(dist), row) selection.
// METRIC: 0 = cosine, 1 = euclidean.
// ---------------------------------------------------------------------------
struct LCand {
	uint64_t key; // total_key(dist)
	double dist;
	uint32_t row;
};

// Block-wide exact top-k selection from an LDS candidate array by ascending
// (key, row). All THREADS threads participate; every cross-thread hand-off is
// ordered by __syncthreads(). Deterministic: selection is by value.
__device__ static void block_select_topk(LCand *buf, int total, LCand *topk,
                                         int k, int *out_n) {
	__shared__ uint64_t red_key[THREADS / 64];
	__shared__ uint32_t red_row[THREADS / 64];
	__shared__ int red_idx[THREADS / 64];
	const int lane = threadIdx.x & 63;
	const int wave = threadIdx.x >> 6;
	int out = total < k ? total : k;
	for (int slot = 0; slot < out; slot++) {
		uint64_t bk = ~0ULL;
		uint32_t br = ~0u;
		int bi = -1;
		for (int i = threadIdx.x; i < total; i += THREADS) {
			uint64_t ck = buf[i].key;
			uint32_t cr = buf[i].row;
			if (ck < bk || (ck == bk && cr < br)) {
				bk = ck;
				br = cr;
				bi = i;
			}
		}
		for (int off = 32; off > 0; off >>= 1) {
			uint64_t ok = __shfl_down(bk, off);
			uint32_t orr = __shfl_down(br, off);
			int oi = __shfl_down(bi, off);
			if (ok < bk || (ok == bk && orr < br)) {
				bk = ok;
				br = orr;
				bi = oi;
			}
		}
		if (lane == 0) {
			red_key[wave] = bk;
			red_row[wave] = br;
			red_idx[wave] = bi;
		}
		__syncthreads();
		if (threadIdx.x == 0) {
			int best = -1;
			uint64_t bbk = ~0ULL;
			uint32_t bbr = ~0u;
			for (int w = 0; w < THREADS / 64; w++) {
				if (red_idx[w] < 0)
					continue;
				if (red_key[w] < bbk ||
				    (red_key[w] == bbk && red_row[w] < bbr)) {
					bbk = red_key[w];
					bbr = red_row[w];
					best = red_idx[w];
				}
			}
			topk[slot] = buf[best];
			buf[best].key = ~0ULL;
			buf[best].row = ~0u;
		}
		__syncthreads();
	}
	*out_n = out;
}

template <int METRIC>
__global__ __launch_bounds__(THREADS) void k_scan(
    const float *__restrict__ cm, const double *__restrict__ norms,
    uint64_t n, uint64_t n_pad, uint32_t d, const float *__restrict__ qg,
    double q_norm, uint32_t k, uint64_t rows_per_block,
    Cand *__restrict__ block_out, const uint64_t *__restrict__ ids) {
	__shared__ float qs[MAX_D];
	__shared__ LCand buf[TILE + MAX_K];
	__shared__ LCand topk[MAX_K];
	__shared__ int cnt;
	__shared__ int topk_n;
	__shared__ uint64_t kth_key;
	__shared__ uint32_t kth_row;

	for (uint32_t i = threadIdx.x; i < d; i += THREADS)
		qs[i] = qg[i];
	if (threadIdx.x == 0) {
		cnt = 0;
		topk_n = 0;
		kth_key = ~0ULL;
		kth_row = ~0u;
	}
	__syncthreads();

	uint64_t row_begin = (uint64_t)blockIdx.x * rows_per_block;
	uint64_t row_end = row_begin + rows_per_block;
	if (row_end > n)
		row_end = n;

	for (uint64_t tile_base = row_begin; tile_base < row_end; tile_base += TILE) {
		uint64_t r0 = tile_base + (uint64_t)threadIdx.x * ROWS_PER_LANE;
		double dist[ROWS_PER_LANE];
		bool valid[ROWS_PER_LANE];
#pragma unroll
		for (int c = 0; c < ROWS_PER_LANE; c++)
			valid[c] = (r0 + c) < row_end;

		if (METRIC == 0) {
			// cosine: ndarray unrolled_dot restatement, per row (float4 = 4 rows)
			float4 p[8];
#pragma unroll
			for (int j = 0; j < 8; j++)
				p[j] = make_float4(0.f, 0.f, 0.f, 0.f);
			uint32_t k8 = 0;
			for (; k8 + 8 <= d; k8 += 8) {
#pragma unroll
				for (uint32_t j = 0; j < 8; j++) {
					const float4 v =
					    *(const float4 *)(cm + (uint64_t)(k8 + j) * n_pad + r0);
					const float qk = qs[k8 + j];
					p[j].x = __fadd_rn(p[j].x, __fmul_rn(v.x, qk));
					p[j].y = __fadd_rn(p[j].y, __fmul_rn(v.y, qk));
					p[j].z = __fadd_rn(p[j].z, __fmul_rn(v.z, qk));
					p[j].w = __fadd_rn(p[j].w, __fmul_rn(v.w, qk));
				}
			}
			float4 sum = make_float4(0.f, 0.f, 0.f, 0.f);
#define COMB(f)                                                                 \
	sum.f = __fadd_rn(sum.f, __fadd_rn(p[0].f, p[4].f));                        \
	sum.f = __fadd_rn(sum.f, __fadd_rn(p[1].f, p[5].f));                        \
	sum.f = __fadd_rn(sum.f, __fadd_rn(p[2].f, p[6].f));                        \
	sum.f = __fadd_rn(sum.f, __fadd_rn(p[3].f, p[7].f));
			COMB(x) COMB(y) COMB(z) COMB(w)
#undef COMB
			for (; k8 < d; k8++) {
				const float4 v = *(const float4 *)(cm + (uint64_t)k8 * n_pad + r0);
				const float qk = qs[k8];
				sum.x = __fadd_rn(sum.x, __fmul_rn(v.x, qk));
				sum.y = __fadd_rn(sum.y, __fmul_rn(v.y, qk));
				sum.z = __fadd_rn(sum.z, __fmul_rn(v.z, qk));
				sum.w = __fadd_rn(sum.w, __fmul_rn(v.w, qk));
			}
			const float s[4] = {sum.x, sum.y, sum.z, sum.w};
#pragma unroll
			for (int c = 0; c < ROWS_PER_LANE; c++) {
				double den = q_norm * norms[r0 + c < n_pad ? r0 + c : 0];
				dist[c] = 1.0 - (double)s[c] / den;
			}
		} else {
			// euclidean: ndarray-stats sq_l2_dist — sequential f32 chain per row
			float4 acc = make_float4(0.f, 0.f, 0.f, 0.f);
			for (uint32_t kk = 0; kk < d; kk++) {
				const float4 v = *(const float4 *)(cm + (uint64_t)kk * n_pad + r0);
				const float qk = qs[kk];
				float dx = v.x - qk, dy = v.y - qk, dz = v.z - qk, dw = v.w - qk;
				acc.x = __fadd_rn(acc.x, __fmul_rn(dx, dx));
				acc.y = __fadd_rn(acc.y, __fmul_rn(dy, dy));
				acc.z = __fadd_rn(acc.z, __fmul_rn(dz, dz));
				acc.w = __fadd_rn(acc.w, __fmul_rn(dw, dw));
			}
			const float s[4] = {acc.x, acc.y, acc.z, acc.w};
#pragma unroll
			for (int c = 0; c < ROWS_PER_LANE; c++)
				dist[c] = sqrt((double)s[c]);
		}

		// threshold check + candidate append (rare path)
		uint64_t kk_key = kth_key;
		uint32_t kk_row = kth_row;
		int full = (topk_n >= (int)k);
#pragma unroll
		for (int c = 0; c < ROWS_PER_LANE; c++) {
			if (!valid[c])
				continue;
			uint64_t key = d_total_key(dist[c]);
			uint32_t row32 = (uint32_t)(r0 + c - 0); // row index within shard
			bool take = !full || key < kk_key || (key == kk_key && row32 < kk_row);
			if (take) {
				int idx = atomicAdd(&cnt, 1);
				buf[idx].key = key;
				buf[idx].dist = dist[c];
				buf[idx].row = row32;
			}
		}
		__syncthreads();

		// merge: all threads select new top-k from {candidates, current topk}
		if (cnt > 0) {
			int m = cnt;
			for (int i = threadIdx.x; i < topk_n; i += THREADS)
				buf[m + i] = topk[i];
			int total = m + topk_n;
			__syncthreads();
			int new_n;
			block_select_topk(buf, total, topk, (int)k, &new_n);
			if (threadIdx.x == 0) {
				topk_n = new_n;
				if (new_n >= (int)k) {
					kth_key = topk[k - 1].key;
					kth_row = topk[k - 1].row;
				}
				cnt = 0;
			}
		}
		__syncthreads();
	}

	// write block winners (pad with +inf)
	if (threadIdx.x < MAX_K && threadIdx.x < k) {
		Cand c;
		if ((int)threadIdx.x < topk_n) {
			c.dist = topk[threadIdx.x].dist;
			c.id = ids[topk[threadIdx.x].row];
		} else {
			c.dist = __longlong_as_double(0x7FF0000000000000LL); // +inf
			c.id = ~0ULL;
		}
		block_out[(uint64_t)blockIdx.x * k + threadIdx.x] = c;
	}
}

// Final merge: one block selects global top-k from nblocks*k candidates.
// block_out is mutable scratch: selected entries are poisoned to +inf
// (same-workgroup global visibility ordered by __syncthreads()).
// Exact k-selection over a candidate slab. Each BLOCK merges its slice
// [per_block*g, per_block*(g+1)) into out[g*k..]: grid 1 with per_block =
// total is the classic single-block merge; a two-level tree (G slices,
// then one block over G*k) cuts the 1954-block scan's merge latency
// (top-k of per-slice top-ks == the global top-k, exactly).
__global__ __launch_bounds__(THREADS) void k_merge(
    Cand *__restrict__ block_out, uint64_t total, uint32_t k,
    uint64_t per_block, Cand *__restrict__ out) {
	__shared__ uint64_t red_key[THREADS / 64];
	__shared__ uint64_t red_id[THREADS / 64];
	__shared__ long red_idx[THREADS / 64];
	const int lane = threadIdx.x & 63;
	const int wave = threadIdx.x >> 6;
	const uint64_t s0 = (uint64_t)blockIdx.x * per_block;
	const uint64_t s1 = s0 + per_block < total ? s0 + per_block : total;
	out += (uint64_t)blockIdx.x * k;
	if (s0 >= total) {
		for (uint32_t i = threadIdx.x; i < k; i += THREADS) {
			out[i].dist = __longlong_as_double(0x7FF0000000000000LL);
			out[i].id = ~0ULL;
		}
		return;
	}
	for (int slot = 0; slot < (int)k; slot++) {
		uint64_t bk = ~0ULL;
		uint64_t bid = ~0ULL;
		long bi = -1;
		for (uint64_t i = s0 + threadIdx.x; i < s1; i += THREADS) {
			uint64_t ck = d_total_key(block_out[i].dist);
			uint64_t cid = block_out[i].id;
			if (ck < bk || (ck == bk && cid < bid)) {
				bk = ck;
				bid = cid;
				bi = (long)i;
			}
		}
		for (int off = 32; off > 0; off >>= 1) {
			uint64_t ok = __shfl_down(bk, off);
			uint64_t oid = __shfl_down(bid, off);
			long oi = __shfl_down(bi, off);
			if (ok < bk || (ok == bk && oid < bid)) {
				bk = ok;
				bid = oid;
				bi = oi;
			}
		}
		if (lane == 0) {
			red_key[wave] = bk;
			red_id[wave] = bid;
			red_idx[wave] = bi;
		}
		__syncthreads();
		if (threadIdx.x == 0) {
			uint64_t bbk = ~0ULL, bbid = ~0ULL;
			long best = -1;
			for (int w = 0; w < THREADS / 64; w++) {
				if (red_idx[w] < 0)
					continue;
				if (red_key[w] < bbk ||
				    (red_key[w] == bbk && red_id[w] < bbid)) {
					bbk = red_key[w];
					bbid = red_id[w];
					best = red_idx[w];
				}
			}
			if (best < 0) { // slice shorter than k: pad
				out[slot].dist =
				    __longlong_as_double(0x7FF0000000000000LL);
				out[slot].id = ~0ULL;
			} else {
				out[slot] = block_out[best];
				block_out[best].dist =
				    __longlong_as_double(0x7FF0000000000000LL);
				block_out[best].id = ~0ULL;
			}
		}
		__syncthreads();
	}
}

// All-distance kernel: one lane per row, feature-major coalesced reads.
// Used for parity dumps, k > MAX_K, and (as the product path proper) the
// six non-headline metrics — each restating the reference's F32 chain
// (vector.rs:206-451) operation-for-operation. `order` = minkowski order;
// `q_mean` = the query's ndarray-mean (pearson, host-precomputed with the
// same unrolled-8 chain).
template <int METRIC>
__global__ void k_all_dists(const float *__restrict__ cm,
                            const double *__restrict__ norms, uint64_t n,
                            uint64_t n_pad, uint32_t d,
                            const float *__restrict__ qg, double q_norm,
                            double order, double q_mean,
                            double *__restrict__ out) {
	uint64_t r = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
	if (r >= n)
		return;
	if (METRIC == 0) {
		float p[8] = {0, 0, 0, 0, 0, 0, 0, 0};
		uint32_t k = 0;
		for (; k + 8 <= d; k += 8) {
#pragma unroll
			for (uint32_t j = 0; j < 8; j++) {
				float x = cm[(uint64_t)(k + j) * n_pad + r];
				p[j] = __fadd_rn(p[j], __fmul_rn(x, qg[k + j]));
			}
		}
		float sum = 0.f;
		sum = __fadd_rn(sum, __fadd_rn(p[0], p[4]));
		sum = __fadd_rn(sum, __fadd_rn(p[1], p[5]));
		sum = __fadd_rn(sum, __fadd_rn(p[2], p[6]));
		sum = __fadd_rn(sum, __fadd_rn(p[3], p[7]));
		for (; k < d; k++)
			sum = __fadd_rn(sum, __fmul_rn(cm[(uint64_t)k * n_pad + r], qg[k]));
		out[r] = 1.0 - (double)sum / (q_norm * norms[r]);
	} else if (METRIC == 1) {
		float acc = 0.f;
		for (uint32_t k = 0; k < d; k++) {
			float diff = cm[(uint64_t)k * n_pad + r] - qg[k];
			acc = __fadd_rn(acc, __fmul_rn(diff, diff));
		}
		out[r] = sqrt((double)acc);
	} else if (METRIC == 2) {
		// manhattan via l1_dist: sequential f32 |a-b| sum (vector.rs:379)
		float acc = 0.f;
		for (uint32_t k = 0; k < d; k++)
			acc = __fadd_rn(acc, fabsf(qg[k] - cm[(uint64_t)k * n_pad + r]));
		out[r] = (double)acc;
	} else if (METRIC == 3) {
		// chebyshev via linf_dist: f32 fmax chain (vector.rs:220-229)
		float m = 0.f;
		for (uint32_t k = 0; k < d; k++)
			m = fmaxf(m, fabsf(qg[k] - cm[(uint64_t)k * n_pad + r]));
		out[r] = (double)m;
	} else if (METRIC == 4) {
		// hamming: count of element-wise != (vector.rs:292-303)
		uint64_t acc = 0;
		for (uint32_t k = 0; k < d; k++)
			if (qg[k] != cm[(uint64_t)k * n_pad + r])
				acc++;
		out[r] = (double)acc;
	} else if (METRIC == 6) {
		// minkowski: f64 |diff|^order sequential sum, final ^(1/order)
		// (vector.rs:389-399)
		double acc = 0.0;
		for (uint32_t k = 0; k < d; k++)
			acc += pow(fabs((double)qg[k] -
			                (double)cm[(uint64_t)k * n_pad + r]),
			           order);
		out[r] = pow(acc, 1.0 / order);
	} else if (METRIC == 7) {
		// pearson similarity used directly as the metric value
		// (vector.rs:413-440): row mean via the unrolled-8 f32 chain
		// (ndarray .mean() = .sum()/len), then sequential f64 moments
		float p[8] = {0, 0, 0, 0, 0, 0, 0, 0};
		uint32_t k = 0;
		for (; k + 8 <= d; k += 8) {
#pragma unroll
			for (uint32_t j = 0; j < 8; j++)
				p[j] = __fadd_rn(p[j], cm[(uint64_t)(k + j) * n_pad + r]);
		}
		float sb = 0.f;
		sb = __fadd_rn(sb, __fadd_rn(__fadd_rn(p[0], p[4]),
		                             __fadd_rn(p[1], p[5])));
		sb = __fadd_rn(sb, __fadd_rn(__fadd_rn(p[2], p[6]),
		                             __fadd_rn(p[3], p[7])));
		for (; k < d; k++)
			sb = __fadd_rn(sb, cm[(uint64_t)k * n_pad + r]);
		double my = (double)(sb / (float)d);
		double sum_xy = 0, sum_x2 = 0, sum_y2 = 0;
		for (uint32_t i = 0; i < d; i++) {
			double dx = (double)qg[i] - q_mean;
			double dy = (double)cm[(uint64_t)i * n_pad + r] - my;
			sum_xy += dx * dy;
			sum_x2 += dx * dx;
			sum_y2 += dy * dy;
		}
		double den = sqrt(sum_x2 * sum_y2);
		out[r] = den == 0.0 ? 0.0 : sum_xy / den;
	}
}

// Jaccard (vector.rs:317-340): bit-pattern sets — |I|/|U| with the
// reference's F32 asymmetry restated as-is. One block per row; two LDS
// open-addressing sets (query uniques, row uniques). The counted
// quantities are order-independent restatements of the reference's
// progressive-insert loop: inter = per row element [bits in unique(q)] or
// [non-first duplicate within the row]; |U| = |unique(q)| + |unique(row)
// \ unique(q)|. d <= 1024 (LDS capacity; the guard is in the caller).
#define JACC_SLOTS 2048
#define JACC_EMPTY 0xFFFFFFFFu
__global__ __launch_bounds__(256) void k_jaccard_dists(
    const float *__restrict__ cm, uint64_t n, uint64_t n_pad, uint32_t d,
    const float *__restrict__ qg, double *__restrict__ out) {
	__shared__ uint32_t setA[JACC_SLOTS];
	__shared__ uint32_t setB[JACC_SLOTS];
	__shared__ uint32_t suA, interS, bNotA;
	__shared__ int sentA, sentB; // the 0xFFFFFFFF bit pattern, if present
	uint64_t r = blockIdx.x;
	if (r >= n)
		return;
	for (uint32_t i = threadIdx.x; i < JACC_SLOTS; i += 256) {
		setA[i] = JACC_EMPTY;
		setB[i] = JACC_EMPTY;
	}
	if (threadIdx.x == 0) {
		suA = 0;
		interS = 0;
		bNotA = 0;
		sentA = 0;
		sentB = 0;
	}
	__syncthreads();
	for (uint32_t i = threadIdx.x; i < d; i += 256) {
		uint32_t bits = __float_as_uint(qg[i]);
		if (bits == JACC_EMPTY) {
			if (atomicOr(&sentA, 1) == 0)
				atomicAdd(&suA, 1);
			continue;
		}
		uint32_t h = (bits * 2654435761u) & (JACC_SLOTS - 1);
		for (;;) {
			uint32_t prev = atomicCAS(&setA[h], JACC_EMPTY, bits);
			if (prev == JACC_EMPTY) {
				atomicAdd(&suA, 1);
				break;
			}
			if (prev == bits)
				break;
			h = (h + 1) & (JACC_SLOTS - 1);
		}
	}
	__syncthreads();
	for (uint32_t i = threadIdx.x; i < d; i += 256) {
		uint32_t bits = __float_as_uint(cm[(uint64_t)i * n_pad + r]);
		bool inA;
		if (bits == JACC_EMPTY) {
			inA = sentA != 0;
		} else {
			inA = false;
			uint32_t h = (bits * 2654435761u) & (JACC_SLOTS - 1);
			for (;;) {
				uint32_t v = setA[h];
				if (v == bits) {
					inA = true;
					break;
				}
				if (v == JACC_EMPTY)
					break;
				h = (h + 1) & (JACC_SLOTS - 1);
			}
		}
		if (inA) {
			atomicAdd(&interS, 1);
			continue;
		}
		if (bits == JACC_EMPTY) {
			if (atomicOr(&sentB, 1) == 0)
				atomicAdd(&bNotA, 1);
			else
				atomicAdd(&interS, 1);
			continue;
		}
		uint32_t h = (bits * 2654435761u) & (JACC_SLOTS - 1);
		for (;;) {
			uint32_t prev = atomicCAS(&setB[h], JACC_EMPTY, bits);
			if (prev == JACC_EMPTY) {
				atomicAdd(&bNotA, 1);
				break;
			}
			if (prev == bits) {
				atomicAdd(&interS, 1);
				break;
			}
			h = (h + 1) & (JACC_SLOTS - 1);
		}
	}
	__syncthreads();
	if (threadIdx.x == 0)
		out[r] = (double)interS / (double)(suA + bNotA);
}

// Gather + distance for HNSW frontier expansion: one block per frontier row,
// coalesced row read (rows gathered from the feature-major store).
template <int METRIC>
__global__ void k_gather_dist(const float *__restrict__ cm,
                              const double *__restrict__ norms, uint64_t n_pad,
                              uint32_t d, const uint32_t *__restrict__ rows,
                              uint32_t nrows, const float *__restrict__ qg,
                              double q_norm, double *__restrict__ out) {
	uint32_t i = blockIdx.x;
	if (i >= nrows)
		return;
	uint32_t r = rows[i];
	// single lane computes the exact per-row chain; other lanes prefetch via L2
	if (threadIdx.x == 0) {
		if (METRIC == 0) {
			float p[8] = {0, 0, 0, 0, 0, 0, 0, 0};
			uint32_t k = 0;
			for (; k + 8 <= d; k += 8)
#pragma unroll
				for (uint32_t j = 0; j < 8; j++)
					p[j] = __fadd_rn(
					    p[j], __fmul_rn(cm[(uint64_t)(k + j) * n_pad + r], qg[k + j]));
			float sum = 0.f;
			sum = __fadd_rn(sum, __fadd_rn(p[0], p[4]));
			sum = __fadd_rn(sum, __fadd_rn(p[1], p[5]));
			sum = __fadd_rn(sum, __fadd_rn(p[2], p[6]));
			sum = __fadd_rn(sum, __fadd_rn(p[3], p[7]));
			for (; k < d; k++)
				sum = __fadd_rn(sum, __fmul_rn(cm[(uint64_t)k * n_pad + r], qg[k]));
			out[i] = 1.0 - (double)sum / (q_norm * norms[r]);
		} else {
			float acc = 0.f;
			for (uint32_t k = 0; k < d; k++) {
				float diff = cm[(uint64_t)k * n_pad + r] - qg[k];
				acc = __fadd_rn(acc, __fmul_rn(diff, diff));
			}
			out[i] = sqrt((double)acc);
		}
	}
}

// ---------------------------------------------------------------------------
// Batched-query path (BASELINE configs[3]) — the genuinely-dense case.
// scores = corpus x Q^T runs as a plain f32 GEMM on MFMA (rocBLAS; the
// feature-major store [d][n_pad] IS the column-major corpus matrix with
// lda = n_pad, so no transform is needed). Selection per query uses a
// monotone f32 approximation of the distance as the key; the survivors
// (k + SLACK per query) are then recomputed with the exact restated chain so
// the final scores are bitwise identical to the single-query path, and
// re-ranked by the exact (total_cmp dist, id) order.
// ---------------------------------------------------------------------------
#define BATCH_SLACK 16

__device__ static inline uint64_t d_total_key32(float x) {
	uint32_t bits = __float_as_uint(x);
	uint32_t k = (bits >> 31) ? ~bits : (bits | 0x80000000u);
	return (uint64_t)k;
}

struct BCand {
	float key; // monotone f32 selection key (smaller = better)
	uint32_t row;
};

__global__ void k_binit(BCand *state, uint64_t total) {
	uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
	if (i < total) {
		state[i].key = __uint_as_float(0x7F800000u); // +inf
		state[i].row = ~0u;
	}
}

// f32 aux for the selection key: cosine -> 1/norm, euclidean -> restated f32
// sum-of-squares per row.
__global__ void k_aux(const float *__restrict__ cm,
                      const double *__restrict__ norms,
                      float *__restrict__ aux, uint64_t n, uint64_t n_pad,
                      uint32_t d, int metric) {
	uint64_t r = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
	if (r >= n)
		return;
	if (metric == 0) {
		aux[r] = (float)(1.0 / norms[r]);
	} else {
		float p[8] = {0, 0, 0, 0, 0, 0, 0, 0};
		uint32_t k = 0;
		for (; k + 8 <= d; k += 8) {
#pragma unroll
			for (uint32_t j = 0; j < 8; j++) {
				float x = cm[(uint64_t)(k + j) * n_pad + r];
				p[j] = __fadd_rn(p[j], __fmul_rn(x, x));
			}
		}
		float acc = 0.0f;
		acc = __fadd_rn(acc,
		                __fadd_rn(__fadd_rn(p[0], p[4]), __fadd_rn(p[1], p[5])));
		acc = __fadd_rn(acc,
		                __fadd_rn(__fadd_rn(p[2], p[6]), __fadd_rn(p[3], p[7])));
		for (; k < d; k++) {
			float x = cm[(uint64_t)k * n_pad + r];
			acc = __fadd_rn(acc, __fmul_rn(x, x));
		}
		aux[r] = acc;
	}
}

// Per-query f64 norms of a row-major Q (restated sumsq chain + f64 sqrt).
__global__ void k_qnorms(const float *__restrict__ Q, uint32_t b, uint32_t d,
                         double *__restrict__ out) {
	uint32_t j = blockIdx.x * blockDim.x + threadIdx.x;
	if (j >= b)
		return;
	const float *a = Q + (uint64_t)j * d;
	float p[8] = {0, 0, 0, 0, 0, 0, 0, 0};
	uint32_t i = 0;
	for (; i + 8 <= d; i += 8)
#pragma unroll
		for (uint32_t t = 0; t < 8; t++)
			p[t] = __fadd_rn(p[t], __fmul_rn(a[i + t], a[i + t]));
	float acc = 0.0f;
	acc = __fadd_rn(acc, __fadd_rn(__fadd_rn(p[0], p[4]), __fadd_rn(p[1], p[5])));
	acc = __fadd_rn(acc, __fadd_rn(__fadd_rn(p[2], p[6]), __fadd_rn(p[3], p[7])));
	for (; i < d; i++)
		acc = __fadd_rn(acc, __fmul_rn(a[i], a[i]));
	out[j] = sqrt((double)acc);
}

// Per-query running top-(kk) update over one GEMM chunk.
// S is [n_chunk x b] column-major (column j = scores of query j, contiguous).
// metric 0 (cosine): key = -(s * inv_norm)   (maximise normalised dot)
// metric 1 (euclid): key = sumsq_row - 2*s   (minimise ||q-r||^2 - qn2)
__global__ __launch_bounds__(THREADS) void k_batch_topk(
    const float *__restrict__ S, const float *__restrict__ aux,
    uint64_t row0, uint32_t n_chunk, uint64_t n, int metric,
    BCand *__restrict__ state, int kk) {
	__shared__ LCand buf[TILE + MAX_K];
	__shared__ LCand topk[MAX_K];
	__shared__ int cnt;
	__shared__ int topk_n;
	__shared__ uint64_t kth_key;
	__shared__ uint32_t kth_row;

	uint32_t j = blockIdx.x;
	const float *col = S + (uint64_t)j * n_chunk;
	BCand *st = state + (uint64_t)j * kk;

	// load current state into LDS topk (keys already total-ordered u64)
	if (threadIdx.x == 0) {
		cnt = 0;
		topk_n = 0;
		kth_key = ~0ULL;
		kth_row = ~0u;
	}
	__syncthreads();
	if (threadIdx.x < (uint32_t)kk) {
		BCand c = st[threadIdx.x];
		if (c.row != ~0u) {
			topk[threadIdx.x].key = d_total_key32(c.key);
			topk[threadIdx.x].dist = (double)c.key;
			topk[threadIdx.x].row = c.row;
			atomicAdd(&topk_n, 1);
		}
	}
	__syncthreads();
	if (threadIdx.x == 0 && topk_n >= kk) {
		kth_key = topk[kk - 1].key;
		kth_row = topk[kk - 1].row;
	}
	__syncthreads();

	for (uint32_t tile = 0; tile < n_chunk; tile += TILE) {
		uint32_t i0 = tile + threadIdx.x * 4;
		uint64_t kk_key = kth_key;
		uint32_t kk_row = kth_row;
		int full = (topk_n >= kk);
		if (i0 + 3 < n_chunk) {
			const float4 s4 = *(const float4 *)(col + i0);
			float sv[4] = {s4.x, s4.y, s4.z, s4.w};
#pragma unroll
			for (int c = 0; c < 4; c++) {
				uint64_t r = row0 + i0 + c;
				if (r >= n)
					continue;
				float key = (metric == 0) ? -(sv[c] * aux[r])
				                          : (aux[r] - 2.0f * sv[c]);
				uint64_t tk = d_total_key32(key);
				uint32_t row32 = (uint32_t)r;
				bool take =
				    !full || tk < kk_key || (tk == kk_key && row32 < kk_row);
				if (take) {
					int idx = atomicAdd(&cnt, 1);
					buf[idx].key = tk;
					buf[idx].dist = (double)key;
					buf[idx].row = row32;
				}
			}
		}
		__syncthreads();
		if (cnt > 0) {
			int m = cnt;
			for (int i = threadIdx.x; i < topk_n; i += THREADS)
				buf[m + i] = topk[i];
			int total = m + topk_n;
			__syncthreads();
			int new_n;
			block_select_topk(buf, total, topk, kk, &new_n);
			if (threadIdx.x == 0) {
				topk_n = new_n;
				if (new_n >= kk) {
					kth_key = topk[kk - 1].key;
					kth_row = topk[kk - 1].row;
				}
				cnt = 0;
			}
		}
		__syncthreads();
	}

	// write back
	if (threadIdx.x < (uint32_t)kk) {
		BCand c;
		if ((int)threadIdx.x < topk_n) {
			c.key = (float)topk[threadIdx.x].dist;
			c.row = topk[threadIdx.x].row;
		} else {
			c.key = __uint_as_float(0x7F800000u);
			c.row = ~0u;
		}
		st[threadIdx.x] = c;
	}
}

// ---------------------------------------------------------------------------
// Fused MFMA scan (the hand-written batch kernel): scores = corpus x Q^T on
// v_mfma_f32_32x32x2_f32 (exact f32 at the 157 TF vector rate), with the
// top-K filter fused into the epilogue — survivors (score key <= the
// query's bootstrapped kth-best threshold) append to a per-query candidate
// buffer; the b x n score matrix NEVER touches HBM (the round-1 rocBLAS +
// k_batch_topk path re-read ~41 GB/step of it). Thresholds bootstrap on the
// first rows via the legacy path, so expected appends/query ~ kk*ln(n/n0).
// Selection semantics are identical to k_batch_topk (exact running top-kk
// by (monotone f32 key, row)): the filter can only drop entries strictly
// worse than the current kk-th, and k_cand_fold recomputes the exact
// running top-kk from the survivors.
// Block tile 128x128 x K=32 LDS stages; 4 waves, each 2x2 subtiles of
// 32x32. A = cm (feature-major = column-major corpus, lda n_pad), B = Q
// row-major [b][d].
// ---------------------------------------------------------------------------
#define FMM_BM 128
#define FMM_BN 128
#define FMM_BK 32
#define FMM_LDS_PAD 4
#define FMM_CAND_CAP 4096u

typedef __attribute__((ext_vector_type(16))) float f32x16;

__global__ __launch_bounds__(256) void k_mfma_scan_topk(
    const float *__restrict__ cm, uint64_t n_pad, uint32_t d,
    const float *__restrict__ Q, uint32_t b, uint64_t row0, uint64_t row1,
    const float *__restrict__ aux, int metric,
    const uint32_t *__restrict__ theta, // per-query kth-best u32 key
    BCand *__restrict__ cand,           // [b][FMM_CAND_CAP]
    uint32_t *__restrict__ cand_cnt) {  // [b]
	__shared__ float As[FMM_BK][FMM_BM + FMM_LDS_PAD];
	__shared__ float Bs[FMM_BK][FMM_BN + FMM_LDS_PAD];
	const uint32_t wave = threadIdx.x >> 6;
	const uint32_t lane = threadIdx.x & 63;
	const uint32_t wm = wave & 1;  // wave row (2x2 wave grid)
	const uint32_t wn = wave >> 1; // wave col
	const uint64_t brow = row0 + (uint64_t)blockIdx.x * FMM_BM;
	const uint32_t bcol = blockIdx.y * FMM_BN;
	f32x16 acc[2][2] = {};

	const uint32_t tid = threadIdx.x;
	// software pipeline: prefetch K-stage t+1 into registers while MFMA
	// consumes stage t from LDS (the naive load->sync->compute cycle
	// measured 73 TF; the guide's untuned same-shape reference is 122)
	// each thread stages 4 float4 of A (k = a_kq + 8r, fixed m4) and 4 of
	// B (j = b_jq + 32r, fixed k4): A has FMM_BK*FMM_BM/4 = 1024 float4
	// per stage over 256 threads, B likewise
	const uint32_t a_m4 = (tid % (FMM_BM / 4)) * 4; // 0..124
	const uint32_t b_k4 = (tid % (FMM_BK / 4)) * 4; // 0..28
	const uint32_t a_kq = tid / (FMM_BM / 4); // 0..7 (k quarter index)
	const uint32_t b_jq = tid / (FMM_BK / 4); // 0..31 (j quarter index)
	float4 pa[4], pb[4];
	auto prefetch = [&](uint32_t k0) {
#pragma unroll
		for (uint32_t r = 0; r < 4; r++) {
			uint32_t k = a_kq + 8 * r;
			pa[r] = *(const float4 *)(cm + (uint64_t)(k0 + k) * n_pad +
			                          brow + a_m4);
			uint32_t j = b_jq + 32 * r;
			pb[r] = *(const float4 *)(Q + (uint64_t)(bcol + j) * d + k0 +
			                          b_k4);
		}
	};
	auto stage_lds = [&]() {
#pragma unroll
		for (uint32_t r = 0; r < 4; r++) {
			uint32_t k = a_kq + 8 * r;
			As[k][a_m4 + 0] = pa[r].x;
			As[k][a_m4 + 1] = pa[r].y;
			As[k][a_m4 + 2] = pa[r].z;
			As[k][a_m4 + 3] = pa[r].w;
			uint32_t j = b_jq + 32 * r;
			Bs[b_k4 + 0][j] = pb[r].x;
			Bs[b_k4 + 1][j] = pb[r].y;
			Bs[b_k4 + 2][j] = pb[r].z;
			Bs[b_k4 + 3][j] = pb[r].w;
		}
	};
	prefetch(0);
	stage_lds();
	__syncthreads();
	for (uint32_t k0 = 0; k0 < d; k0 += FMM_BK) {
		if (k0 + FMM_BK < d)
			prefetch(k0 + FMM_BK); // in flight during the MFMA block
#pragma unroll
		for (uint32_t kk = 0; kk < FMM_BK; kk += 2) {
			// operand map (32x32x2): lane l holds A[i=l&31][k=l>>5],
			// B[k=l>>5][j=l&31]
			const uint32_t ai = lane & 31, ak = lane >> 5;
			float a0 = As[kk + ak][wm * 64 + ai];
			float a1 = As[kk + ak][wm * 64 + 32 + ai];
			float b0 = Bs[kk + ak][wn * 64 + ai];
			float b1 = Bs[kk + ak][wn * 64 + 32 + ai];
			acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0,
			                                                 acc[0][0], 0, 0, 0);
			acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0,
			                                                 acc[1][0], 0, 0, 0);
			acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1,
			                                                 acc[0][1], 0, 0, 0);
			acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1,
			                                                 acc[1][1], 0, 0, 0);
		}
		__syncthreads();
		if (k0 + FMM_BK < d) {
			stage_lds();
			__syncthreads();
		}
	}

	// epilogue: C/D map for 32x32 shapes — col = lane&31,
	// row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
	const uint32_t ccol = lane & 31;
	const uint32_t crow_base = 4 * (lane >> 5);
#pragma unroll
	for (uint32_t sn = 0; sn < 2; sn++) {
		uint32_t j = bcol + wn * 64 + sn * 32 + ccol;
		if (j >= b)
			continue;
		uint32_t th = theta[j];
#pragma unroll
		for (uint32_t sm = 0; sm < 2; sm++) {
#pragma unroll
			for (uint32_t rg = 0; rg < 16; rg++) {
				uint32_t rsub =
				    (rg & 3) + 8 * (rg >> 2) + crow_base;
				uint64_t row =
				    brow + wm * 64 + sm * 32 + rsub;
				if (row >= row1)
					continue;
				float s = acc[sm][sn][rg];
				float key = (metric == 0) ? -(s * aux[row])
				                          : (aux[row] - 2.0f * s);
				uint32_t tk = (uint32_t)d_total_key32(key);
				if (tk <= th) {
					uint32_t idx = atomicAdd(&cand_cnt[j], 1u);
					if (idx < FMM_CAND_CAP) {
						cand[(uint64_t)j * FMM_CAND_CAP + idx] =
						    BCand{key, (uint32_t)row};
					}
				}
			}
		}
	}
}

// Per-query thresholds from the bootstrapped running top-kk state.
__global__ void k_theta_init(const BCand *__restrict__ state, int kk,
                             uint32_t b, uint32_t *__restrict__ theta) {
	uint32_t j = blockIdx.x * blockDim.x + threadIdx.x;
	if (j >= b)
		return;
	BCand kth = state[(uint64_t)j * kk + (kk - 1)];
	theta[j] = kth.row == ~0u ? 0xFFFFFFFFu
	                          : (uint32_t)d_total_key32(kth.key);
}

// Fold each query's candidate buffer into its running top-kk state —
// identical comparator and tile loop as k_batch_topk, reading precomputed
// (key, row) pairs instead of a score matrix.
__global__ __launch_bounds__(THREADS) void k_cand_fold(
    const BCand *__restrict__ cand, const uint32_t *__restrict__ cand_cnt,
    BCand *__restrict__ state, int kk) {
	__shared__ LCand buf[TILE + MAX_K];
	__shared__ LCand topk[MAX_K];
	__shared__ int cnt;
	__shared__ int topk_n;
	__shared__ uint64_t kth_key;
	__shared__ uint32_t kth_row;
	uint32_t j = blockIdx.x;
	uint32_t m_in = cand_cnt[j];
	if (m_in > FMM_CAND_CAP)
		m_in = FMM_CAND_CAP; // overflow: host reruns via the legacy path
	const BCand *cj = cand + (uint64_t)j * FMM_CAND_CAP;
	BCand *st = state + (uint64_t)j * kk;
	if (threadIdx.x == 0) {
		cnt = 0;
		topk_n = 0;
		kth_key = ~0ULL;
		kth_row = ~0u;
	}
	__syncthreads();
	if (threadIdx.x < (uint32_t)kk) {
		BCand c = st[threadIdx.x];
		if (c.row != ~0u) {
			topk[threadIdx.x].key = d_total_key32(c.key);
			topk[threadIdx.x].dist = (double)c.key;
			topk[threadIdx.x].row = c.row;
			atomicAdd(&topk_n, 1);
		}
	}
	__syncthreads();
	if (threadIdx.x == 0 && topk_n >= kk) {
		kth_key = topk[kk - 1].key;
		kth_row = topk[kk - 1].row;
	}
	__syncthreads();
	for (uint32_t tile = 0; tile < m_in; tile += TILE) {
		uint32_t i = tile + threadIdx.x;
		uint64_t kk_key = kth_key;
		uint32_t kk_row = kth_row;
		int full = (topk_n >= kk);
		if (i < m_in) {
			BCand c = cj[i];
			uint64_t tk = d_total_key32(c.key);
			bool take = !full || tk < kk_key ||
			            (tk == kk_key && c.row < kk_row);
			if (take) {
				int idx = atomicAdd(&cnt, 1);
				buf[idx].key = tk;
				buf[idx].dist = (double)c.key;
				buf[idx].row = c.row;
			}
		}
		__syncthreads();
		if (cnt > 0) {
			int m = cnt;
			for (int i2 = threadIdx.x; i2 < topk_n; i2 += THREADS)
				buf[m + i2] = topk[i2];
			int total = m + topk_n;
			__syncthreads();
			int new_n;
			block_select_topk(buf, total, topk, kk, &new_n);
			if (threadIdx.x == 0) {
				topk_n = new_n;
				if (new_n >= kk) {
					kth_key = topk[kk - 1].key;
					kth_row = topk[kk - 1].row;
				}
				cnt = 0;
			}
		}
		__syncthreads();
	}
	if (threadIdx.x < (uint32_t)kk) {
		BCand c;
		if ((int)threadIdx.x < topk_n) {
			c.key = (float)topk[threadIdx.x].dist;
			c.row = topk[threadIdx.x].row;
		} else {
			c.key = __uint_as_float(0x7F800000u);
			c.row = ~0u;
		}
		st[threadIdx.x] = c;
	}
}

// Exact recompute of each surviving candidate with the restated per-row
// chain (bitwise identical to the single-query scan path).
__global__ void k_batch_exact(const float *__restrict__ cm,
                              const double *__restrict__ norms,
                              uint64_t n_pad, uint32_t d,
                              const float *__restrict__ Q,
                              const double *__restrict__ qnorms, int metric,
                              const BCand *__restrict__ state, int kk,
                              double *__restrict__ out) {
	uint32_t j = blockIdx.x;  // query
	uint32_t c = blockIdx.y;  // candidate slot
	if (threadIdx.x != 0)
		return;
	BCand cand = state[(uint64_t)j * kk + c];
	double *o = out + (uint64_t)j * kk + c;
	if (cand.row == ~0u) {
		*o = __longlong_as_double(0x7FF0000000000000LL);
		return;
	}
	const float *q = Q + (uint64_t)j * d;
	uint64_t r = cand.row;
	if (metric == 0) {
		float p[8] = {0, 0, 0, 0, 0, 0, 0, 0};
		uint32_t k = 0;
		for (; k + 8 <= d; k += 8)
#pragma unroll
			for (uint32_t t = 0; t < 8; t++)
				p[t] = __fadd_rn(
				    p[t], __fmul_rn(cm[(uint64_t)(k + t) * n_pad + r], q[k + t]));
		float sum = 0.f;
		sum = __fadd_rn(sum, __fadd_rn(p[0], p[4]));
		sum = __fadd_rn(sum, __fadd_rn(p[1], p[5]));
		sum = __fadd_rn(sum, __fadd_rn(p[2], p[6]));
		sum = __fadd_rn(sum, __fadd_rn(p[3], p[7]));
		for (; k < d; k++)
			sum = __fadd_rn(sum, __fmul_rn(cm[(uint64_t)k * n_pad + r], q[k]));
		*o = 1.0 - (double)sum / (qnorms[j] * norms[r]);
	} else {
		float acc = 0.f;
		for (uint32_t k = 0; k < d; k++) {
			float diff = cm[(uint64_t)k * n_pad + r] - q[k];
			acc = __fadd_rn(acc, __fmul_rn(diff, diff));
		}
		*o = sqrt((double)acc);
	}
}

// ---------------------------------------------------------------------------
// Persistent HNSW ef-search kernel: ONE query per 64-thread workgroup, the
// entire best-first loop (layer.rs:184-223) inside one launch. Queues live in
// LDS as append-only arrays with wave-parallel (key,seq) min/max scans —
// exactly the DoublePriorityQueue semantics (knn.rs:15-123: total_cmp order,
// FIFO within equal distance, pop_last = latest of the max key). Neighbour
// distances: one LANE per row over the ROW-MAJOR vector copy with the
// restated per-row chain (bit-identical to the oracle / single-query path).
// Visited set: per-query global bitset.
// ---------------------------------------------------------------------------
#define HQ_CAND_CAP 2048
#define HQ_EF_CAP 512
#define HQ_FLAG_OVERFLOW 1u

// PADDED=0: offsets/edges are the scrubbed layer-0 CSR (search path after
// finalize). PADDED=1: `offsets` is a per-node degree array and `edges` a
// padded [n][stride] adjacency — the GPU snapshot build's delta-updatable
// graph (sdbv_hnsw_insert_batch_snapshot_gpu), where per-chunk edge
// changes scatter into rows instead of re-laying-out a CSR.
// `ep_off` selects the seeding mode: NULL = one entry point per query
// (ep_rows[qid]); non-NULL = the insert path's multi-ep seeding — query
// qid seeds from ep_rows/ep_dists[ep_off[qid] .. ep_off[qid+1]), already
// in (dist total_cmp, seq) order (the previous layer's w, layer.rs:342).
template <int PADDED>
__global__ __launch_bounds__(64) void k_hnsw_search(
    const float *__restrict__ rm, const double *__restrict__ norms,
    uint32_t d, int metric, const uint32_t *__restrict__ offsets,
    const uint32_t *__restrict__ edges, uint32_t stride,
    const float *__restrict__ Q,
    const double *__restrict__ qnorms, const uint32_t *__restrict__ ep_rows,
    const double *__restrict__ ep_dists, const uint32_t *__restrict__ ep_off,
    uint32_t *__restrict__ visited,
    uint64_t vwords_per_q, uint32_t k, uint32_t ef,
    uint32_t *__restrict__ out_rows, double *__restrict__ out_dists,
    uint32_t *__restrict__ out_cnt, uint32_t *__restrict__ out_flags) {
	__shared__ uint64_t c_key[HQ_CAND_CAP];
	__shared__ uint32_t c_seq[HQ_CAND_CAP];
	__shared__ uint32_t c_row[HQ_CAND_CAP];
	__shared__ uint64_t w_key[HQ_EF_CAP];
	__shared__ uint32_t w_seq[HQ_EF_CAP];
	__shared__ uint32_t w_row[HQ_EF_CAP];

	const uint32_t qid = blockIdx.x;
	const int lane = threadIdx.x;
	const float *q = Q + (uint64_t)qid * d;
	const double qn = qnorms[qid];
	uint32_t *vis = visited + (uint64_t)qid * vwords_per_q;

	// state kept wave-uniform in registers (updated by every lane alike)
	uint32_t c_cnt = 0;   // append cursor of candidates
	uint32_t w_cnt = 0;
	uint32_t seq = 0;
	uint32_t flags = 0;
	uint64_t fq_key = ~0ULL; // max key in w (or +max when w not full)
	uint32_t fq_idx = 0;

	// seed with the entry point(s) (search_single layer.rs:76-90; insert's
	// search_multi seeds the whole previous-layer w, layer.rs:342-358)
	{
		uint32_t e0 = ep_off ? ep_off[qid] : qid;
		uint32_t e1 = ep_off ? ep_off[qid + 1] : qid + 1;
		uint32_t ne = e1 - e0;
		for (uint32_t i = lane; i < ne; i += 64) {
			uint64_t ek = d_total_key(ep_dists[e0 + i]);
			uint32_t er = ep_rows[e0 + i];
			c_key[i] = ek;
			c_seq[i] = i;
			c_row[i] = er;
			w_key[i] = ek;
			w_seq[i] = i;
			w_row[i] = er;
			atomicOr(&vis[er >> 5], 1u << (er & 31));
		}
		c_cnt = ne;
		w_cnt = ne;
		seq = ne;
		// eps arrive sorted ascending (key, seq): the max is the last
		fq_key = d_total_key(ep_dists[e1 - 1]);
		fq_idx = ne - 1;
	}
	__syncthreads();

	auto wave_min_cand = [&](uint64_t *bk, uint32_t *bs, int *bi) {
		uint64_t mk = ~0ULL;
		uint32_t ms = ~0u;
		int mi = -1;
		for (uint32_t i = lane; i < c_cnt; i += 64) {
			uint64_t kk = c_key[i];
			uint32_t ss = c_seq[i];
			if (kk < mk || (kk == mk && ss < ms)) {
				mk = kk;
				ms = ss;
				mi = (int)i;
			}
		}
		for (int off = 32; off > 0; off >>= 1) {
			uint64_t ok = __shfl_down(mk, off);
			uint32_t os = __shfl_down(ms, off);
			int oi = __shfl_down(mi, off);
			if (ok < mk || (ok == mk && os < ms)) {
				mk = ok;
				ms = os;
				mi = oi;
			}
		}
		*bk = __shfl(mk, 0);
		*bs = __shfl(ms, 0);
		*bi = __shfl(mi, 0);
	};
	auto wave_max_w = [&](uint64_t *bk, uint32_t *bi) {
		// max by (key, seq): latest seq among the max key
		uint64_t mk = 0;
		uint32_t ms = 0;
		int mi = 0;
		for (uint32_t i = lane; i < w_cnt; i += 64) {
			uint64_t kk = w_key[i];
			uint32_t ss = w_seq[i];
			if (kk > mk || (kk == mk && ss > ms)) {
				mk = kk;
				ms = ss;
				mi = (int)i;
			}
		}
		for (int off = 32; off > 0; off >>= 1) {
			uint64_t ok = __shfl_down(mk, off);
			uint32_t os = __shfl_down(ms, off);
			int oi = __shfl_down(mi, off);
			if (ok > mk || (ok == mk && os > ms)) {
				mk = ok;
				ms = os;
				mi = oi;
			}
		}
		*bk = __shfl(mk, 0);
		*bi = __shfl(mi, 0);
	};

	// main best-first loop
	for (;;) {
		uint64_t ck;
		uint32_t cs;
		int ci;
		wave_min_cand(&ck, &cs, &ci);
		if (ci < 0)
			break;
		// cq_dist > fq_dist -> stop (strict, distance only; fq = max of w
		// from the very first element, layer.rs:184)
		if (ck > fq_key)
			break;
		uint32_t doc = c_row[ci];
		if (lane == 0)
			c_key[ci] = ~0ULL; // tombstone
		__syncthreads();

		uint32_t e0, e1;
		if (PADDED) {
			e0 = doc * stride;
			e1 = e0 + offsets[doc]; // offsets = per-node degree
		} else {
			e0 = offsets[doc];
			e1 = offsets[doc + 1];
		}
		for (uint32_t base = e0; base < e1; base += 64) {
			uint32_t my_e = ~0u;
			double my_d = 0.0;
			bool mine = false;
			uint32_t idx = base + lane;
			if (idx < e1) {
				uint32_t e = edges[idx];
				uint32_t old = atomicOr(&vis[e >> 5], 1u << (e & 31));
				if (!(old & (1u << (e & 31)))) {
					mine = true;
					my_e = e;
					// restated per-row chain from the row-major copy
					const float *row = rm + (uint64_t)e * d;
					if (metric == 0) {
						float p[8] = {0, 0, 0, 0, 0, 0, 0, 0};
						uint32_t kk = 0;
						for (; kk + 8 <= d; kk += 8)
#pragma unroll
							for (uint32_t t = 0; t < 8; t++)
								p[t] = __fadd_rn(
								    p[t], __fmul_rn(row[kk + t], q[kk + t]));
						float sum = 0.f;
						sum = __fadd_rn(sum, __fadd_rn(p[0], p[4]));
						sum = __fadd_rn(sum, __fadd_rn(p[1], p[5]));
						sum = __fadd_rn(sum, __fadd_rn(p[2], p[6]));
						sum = __fadd_rn(sum, __fadd_rn(p[3], p[7]));
						for (; kk < d; kk++)
							sum = __fadd_rn(sum,
							                __fmul_rn(row[kk], q[kk]));
						my_d = 1.0 - (double)sum / (qn * norms[e]);
					} else {
						float acc = 0.f;
						for (uint32_t kk = 0; kk < d; kk++) {
							float diff = row[kk] - q[kk];
							acc = __fadd_rn(acc, __fmul_rn(diff, diff));
						}
						my_d = sqrt((double)acc);
					}
				}
			}
			// serial accept/update in edge order (layer.rs:195-218)
			uint32_t nlanes = (e1 - base) < 64 ? (e1 - base) : 64;
			for (uint32_t i = 0; i < nlanes; i++) {
				int src = (int)i;
				bool m = __shfl(mine ? 1 : 0, src) != 0;
				if (!m)
					continue;
				uint32_t erow = __shfl(my_e, src);
				double ed = __shfl(my_d, src);
				uint64_t ekey = d_total_key(ed);
				// e_dist < fq_dist || w.len < ef  (strict distance compare)
				if (!(w_cnt < ef || ekey < fq_key))
					continue;
				// candidates.push
				if (c_cnt < HQ_CAND_CAP) {
					if (lane == 0) {
						c_key[c_cnt] = ekey;
						c_seq[c_cnt] = seq;
						c_row[c_cnt] = erow;
					}
					c_cnt++;
				} else {
					flags |= HQ_FLAG_OVERFLOW;
				}
				__syncthreads();
				// w.push + (w.len > ef ? pop_last) as a conditional replace
				bool replaced = false;
				if (w_cnt < ef) {
					if (lane == 0) {
						w_key[w_cnt] = ekey;
						w_seq[w_cnt] = seq;
						w_row[w_cnt] = erow;
					}
					// incremental max: the new entry has the largest seq, so
					// ekey >= fq_key makes it the (key,seq) max
					if (ekey >= fq_key) {
						fq_key = ekey;
						fq_idx = w_cnt;
					}
					w_cnt++;
				} else {
					// new entry has the LARGEST seq: if its key >= max key it
					// would be popped right back (pop_last = latest of the
					// max) -> net no-op; else it replaces the current max
					if (ekey < fq_key) {
						if (lane == 0) {
							w_key[fq_idx] = ekey;
							w_seq[fq_idx] = seq;
							w_row[fq_idx] = erow;
						}
						replaced = true;
					}
				}
				seq++;
				__syncthreads();
				if (replaced)
					wave_max_w(&fq_key, &fq_idx);
			}
		}
	}

	// emit to_vec_limit(k): ascending (key, seq) = (dist, FIFO) (knn.rs:92-104)
	uint32_t out_m = w_cnt < k ? w_cnt : k;
	for (uint32_t slot = 0; slot < out_m; slot++) {
		uint64_t mk = ~0ULL;
		uint32_t ms = ~0u;
		int mi = -1;
		for (uint32_t i = lane; i < w_cnt; i += 64) {
			uint64_t kk = w_key[i];
			uint32_t ss = w_seq[i];
			if (kk < mk || (kk == mk && ss < ms)) {
				mk = kk;
				ms = ss;
				mi = (int)i;
			}
		}
		for (int off = 32; off > 0; off >>= 1) {
			uint64_t ok = __shfl_down(mk, off);
			uint32_t os = __shfl_down(ms, off);
			int oi = __shfl_down(mi, off);
			if (ok < mk || (ok == mk && os < ms)) {
				mk = ok;
				ms = os;
				mi = oi;
			}
		}
		mk = __shfl(mk, 0);
		mi = __shfl(mi, 0);
		if (lane == 0) {
			out_rows[(uint64_t)qid * k + slot] = w_row[mi];
			// decode distance from the total_cmp key (invertible)
			uint64_t bits =
			    (mk >> 63) ? (mk & ~0x8000000000000000ULL) : ~mk;
			out_dists[(uint64_t)qid * k + slot] =
			    __longlong_as_double(bits);
			w_key[mi] = ~0ULL;
			w_seq[mi] = ~0u;
		}
		__syncthreads();
	}
	if (lane == 0) {
		out_cnt[qid] = out_m;
		out_flags[qid] = flags;
	}
}

// Scatter per-chunk adjacency deltas into the padded device graph: upd_ids
// names the dirty nodes, upd_deg their new degrees, upd_edges their new
// edge rows (cnt x stride, host-packed). One thread per slot.
__global__ void k_adj_scatter(const uint32_t *__restrict__ upd_ids,
                              const uint32_t *__restrict__ upd_deg,
                              const uint32_t *__restrict__ upd_edges,
                              uint32_t cnt, uint32_t stride,
                              uint32_t *__restrict__ adj,
                              uint32_t *__restrict__ deg) {
	uint64_t t = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
	if (t >= (uint64_t)cnt * stride)
		return;
	uint32_t i = (uint32_t)(t / stride);
	uint32_t s = (uint32_t)(t % stride);
	uint32_t node = upd_ids[i];
	if (s == 0)
		deg[node] = upd_deg[i];
	if (s < upd_deg[i])
		adj[(uint64_t)node * stride + s] = upd_edges[t];
}

// ---------------------------------------------------------------------------
// Batched-apply kernels for the GPU snapshot build: the host apply phase is
// RAM-bound on pair distances (select's is_closer + backlink prunes read
// ~100 MB of random rows per insert), so the pair matrices and the
// heuristic select itself run on the device; the host keeps only the
// (cheap, exact) edge bookkeeping. Distances use the restated chain, so a
// host twin with the same batched schedule is bit-identical.
// ---------------------------------------------------------------------------

// One block per list. lists = concatenated [focus, c_0, c_1, ...] row ids
// (loff[i]..loff[i+1]); mats = per-list len^2 f64 distance matrix at
// moff[i] (row-major, M[a][b] = dist(list[a], list[b])).
__global__ __launch_bounds__(256) void k_pair_mats(
    const float *__restrict__ rm, const double *__restrict__ norms,
    uint32_t d, int metric, const uint32_t *__restrict__ lists,
    const uint32_t *__restrict__ loff, const uint64_t *__restrict__ moff,
    double *__restrict__ mats) {
	uint32_t li = blockIdx.x;
	uint32_t o0 = loff[li], o1 = loff[li + 1];
	uint32_t len = o1 - o0;
	double *M = mats + moff[li];
	for (uint32_t p = threadIdx.x; p < len * len; p += 256) {
		uint32_t a = p / len, bb = p % len;
		if (bb < a)
			continue; // symmetric: fill upper, mirror below
		uint32_t ra = lists[o0 + a], rb = lists[o0 + bb];
		const float *va = rm + (uint64_t)ra * d;
		const float *vb = rm + (uint64_t)rb * d;
		double dist;
		if (metric == 0) {
			float pp[8] = {0, 0, 0, 0, 0, 0, 0, 0};
			uint32_t k = 0;
			for (; k + 8 <= d; k += 8)
#pragma unroll
				for (uint32_t t = 0; t < 8; t++)
					pp[t] = __fadd_rn(pp[t],
					                  __fmul_rn(va[k + t], vb[k + t]));
			float sum = 0.f;
			sum = __fadd_rn(sum, __fadd_rn(pp[0], pp[4]));
			sum = __fadd_rn(sum, __fadd_rn(pp[1], pp[5]));
			sum = __fadd_rn(sum, __fadd_rn(pp[2], pp[6]));
			sum = __fadd_rn(sum, __fadd_rn(pp[3], pp[7]));
			for (; k < d; k++)
				sum = __fadd_rn(sum, __fmul_rn(va[k], vb[k]));
			dist = 1.0 - (double)sum / (norms[ra] * norms[rb]);
		} else {
			float acc = 0.f;
			for (uint32_t k = 0; k < d; k++) {
				float diff = va[k] - vb[k];
				acc = __fadd_rn(acc, __fmul_rn(diff, diff));
			}
			dist = sqrt((double)acc);
		}
		M[(uint64_t)a * len + bb] = dist;
		if (a != bb)
			M[(uint64_t)bb * len + a] = dist;
	}
}

// Heuristic neighbour select on the device (heuristic.rs:35-116 without
// extend, + keep :92-112; is_closer :193-216): one 64-lane block per list.
// Candidates = list[1..len), keys = M[0][i], popped in (total_cmp key,
// insertion seq) order — reproduced by a stable sort on (key, index).
// out_sel[i][m_max] gets the selected ROW ids, out_cnt[i] the count.
#define HSEL_CAP 640 // 1 + HQ_EF_CAP + headroom; m_max <= 64
__global__ __launch_bounds__(64) void k_heur_select(
    const uint32_t *__restrict__ lists, const uint32_t *__restrict__ loff,
    const uint64_t *__restrict__ moff, const double *__restrict__ mats,
    uint32_t m_max, int keep, uint32_t *__restrict__ out_sel,
    uint32_t *__restrict__ out_cnt) {
	__shared__ uint64_t skey[HSEL_CAP];
	__shared__ uint32_t sidx[HSEL_CAP];
	__shared__ uint32_t res[64];
	__shared__ uint32_t pruned[HSEL_CAP];
	uint32_t li = blockIdx.x;
	uint32_t o0 = loff[li], o1 = loff[li + 1];
	uint32_t len = o1 - o0;
	uint32_t n = len - 1; // candidates
	const double *M = mats + moff[li];
	const int lane = threadIdx.x;
	uint32_t *out = out_sel + (uint64_t)li * m_max;
	for (uint32_t i = lane; i < n; i += 64) {
		skey[i] = d_total_key(M[1 + i]);
		sidx[i] = i;
	}
	__syncthreads();
	// odd-even transposition sort on (key, idx) — stable order == the
	// host PQ's (total_cmp, push seq)
	for (uint32_t phase = 0; phase < n; phase++) {
		uint32_t start = phase & 1;
		for (uint32_t i = start + 2 * lane; i + 1 < n; i += 128) {
			uint64_t k0 = skey[i], k1 = skey[i + 1];
			uint32_t i0 = sidx[i], i1 = sidx[i + 1];
			if (k1 < k0 || (k1 == k0 && i1 < i0)) {
				skey[i] = k1;
				skey[i + 1] = k0;
				sidx[i] = i1;
				sidx[i + 1] = i0;
			}
		}
		__syncthreads();
	}
	if (n <= m_max) {
		for (uint32_t i = lane; i < n; i += 64)
			out[i] = lists[o0 + 1 + sidx[i]];
		if (lane == 0)
			out_cnt[li] = n;
		return;
	}
	// serial heuristic loop; is_closer's res scan is lane-parallel
	__shared__ int res_n, pruned_n, done;
	if (lane == 0) {
		res_n = 0;
		pruned_n = 0;
		done = 0;
	}
	__syncthreads();
	for (uint32_t i = 0; i < n; i++) {
		if (done)
			break;
		uint32_t ci = sidx[i];          // list index - 1
		double ed = M[1 + ci];          // dist(focus, cand)
		// is_closer: fails if ed > dist(cand, res_r) for ANY r
		int bad = 0;
		if (lane < res_n) {
			double dr = M[(uint64_t)(1 + ci) * len + (1 + res[lane])];
			bad = ed > dr;
		}
		uint64_t anybad = __ballot(bad);
		if (lane == 0) {
			if (!anybad) {
				res[res_n++] = ci;
				if ((uint32_t)res_n == m_max)
					done = 1;
			} else if (keep) {
				pruned[pruned_n++] = ci;
			}
		}
		__syncthreads();
	}
	if (keep) {
		__syncthreads();
		if (lane == 0) {
			int nmore = (int)m_max - res_n;
			for (int i = 0; i < nmore && i < pruned_n; i++)
				res[res_n++] = pruned[i];
		}
		__syncthreads();
	}
	for (int i = lane; i < res_n; i += 64)
		out[i] = lists[o0 + 1 + res[i]];
	if (lane == 0)
		out_cnt[li] = (uint32_t)res_n;
}

// ---------------------------------------------------------------------------
// Host-side helpers (restated reference arithmetic for the query's own norm —
// must be bit-identical to oracle orc_sumsq_f32; covered by tests/)
// ---------------------------------------------------------------------------
static float h_sumsq_f32(const float *a, uint32_t d) {
	float p[8] = {0, 0, 0, 0, 0, 0, 0, 0};
	uint32_t i = 0;
	for (; i + 8 <= d; i += 8)
		for (uint32_t j = 0; j < 8; j++)
			p[j] += a[i + j] * a[i + j];
	float acc = 0;
	acc += ((p[0] + p[4]) + (p[1] + p[5]));
	acc += ((p[2] + p[6]) + (p[3] + p[7]));
	for (; i < d; i++)
		acc += a[i] * a[i];
	return acc;
}

// ndarray .mean() = unrolled-8 .sum() / len in f32, then widened (the
// pearson query mean, vector.rs:418-421 via ndarray; same chain as the
// oracle's restatement).
static double h_mean_f32(const float *a, uint32_t d) {
	float p[8] = {0, 0, 0, 0, 0, 0, 0, 0};
	uint32_t i = 0;
	for (; i + 8 <= d; i += 8)
		for (uint32_t j = 0; j < 8; j++)
			p[j] += a[i + j];
	float s = 0;
	s += ((p[0] + p[4]) + (p[1] + p[5]));
	s += ((p[2] + p[6]) + (p[3] + p[7]));
	for (; i < d; i++)
		s += a[i];
	return (double)(s / (float)d);
}

static void free_table(Table &t) {
	if (t.cm)
		(void)hipFree(t.cm);
	if (t.norms)
		(void)hipFree(t.norms);
	if (t.aux)
		(void)hipFree(t.aux);
	if (t.ids_dev)
		(void)hipFree(t.ids_dev);
	t = Table{};
}

// ---------------------------------------------------------------------------
// C-ABI implementation
// ---------------------------------------------------------------------------
extern "C" {

int sdbv_init(int device, sdbv_ctx **out) {
	auto *ctx = new sdbv_ctx();
	if (device >= 0) {
		if (hipSetDevice(device) != hipSuccess) {
			delete ctx;
			return SDBV_ERR_HIP;
		}
		ctx->device = device;
	} else {
		(void)hipGetDevice(&ctx->device);
	}
	if (hipStreamCreate(&ctx->stream) != hipSuccess) {
		delete ctx;
		return SDBV_ERR_HIP;
	}
	(void)hipEventCreate(&ctx->ev0);
	(void)hipEventCreate(&ctx->ev1);
	(void)hipEventCreate(&ctx->ev2);
	*out = ctx;
	return SDBV_OK;
}

void sdbv_shutdown(sdbv_ctx *ctx) {
	if (!ctx)
		return;
	for (auto &kv : ctx->tables)
		free_table(kv.second);
	if (ctx->block_out)
		(void)hipFree(ctx->block_out);
	if (ctx->final_out)
		(void)hipFree(ctx->final_out);
	if (ctx->merge_tmp)
		(void)hipFree(ctx->merge_tmp);
	if (ctx->q_dev)
		(void)hipFree(ctx->q_dev);
	for (void *p : {(void *)ctx->S, ctx->bstate, (void *)ctx->Q_dev,
	                (void *)ctx->qnorms, (void *)ctx->bdists,
	                (void *)ctx->btheta, (void *)ctx->bcand,
	                (void *)ctx->bcand_cnt})
		if (p)
			(void)hipFree(p);
	if (ctx->blas)
		(void)rocblas_destroy_handle(ctx->blas);
	(void)hipEventDestroy(ctx->ev0);
	(void)hipEventDestroy(ctx->ev1);
	(void)hipEventDestroy(ctx->ev2);
	(void)hipStreamDestroy(ctx->stream);
	delete ctx;
}

const char *sdbv_last_error(sdbv_ctx *ctx) { return ctx ? ctx->err.c_str() : ""; }

int sdbv_get_stats(sdbv_ctx *ctx, sdbv_stats *out) {
	if (!ctx || !out)
		return SDBV_ERR_BAD_ARG;
	*out = ctx->stats;
	return SDBV_OK;
}

static int stage_common(sdbv_ctx *ctx, uint64_t table, uint64_t n, uint32_t d,
                        uint8_t metric, Table **out) {
	if (d == 0 || d > MAX_D || (d % 4) != 0)
		return SDBV_ERR_BAD_ARG; // float4 path needs d%4==0 in this revision
	if (metric > SDBV_METRIC_PEARSON)
		return SDBV_ERR_UNSUPPORTED;
	if (metric == SDBV_METRIC_JACCARD && d > 1024)
		return SDBV_ERR_UNSUPPORTED; // LDS set capacity (JACC_SLOTS)
	auto it = ctx->tables.find(table);
	if (it != ctx->tables.end()) {
		free_table(it->second);
		ctx->tables.erase(it);
	}
	Table t;
	t.n = n;
	t.n_pad = ((n + TILE - 1) / TILE) * TILE;
	t.d = d;
	t.metric = metric;
	uint64_t cm_bytes = (uint64_t)d * t.n_pad * sizeof(float);
	HIP_CHECK(ctx, hipMalloc(&t.cm, cm_bytes));
	t.bytes = cm_bytes;
	if (metric == SDBV_METRIC_COSINE) {
		HIP_CHECK(ctx, hipMalloc(&t.norms, t.n_pad * sizeof(double)));
		t.bytes += t.n_pad * sizeof(double);
	}
	HIP_CHECK(ctx, hipMalloc(&t.ids_dev, t.n_pad * sizeof(uint64_t)));
	t.bytes += t.n_pad * sizeof(uint64_t);
	ctx->tables[table] = t;
	*out = &ctx->tables[table];
	return SDBV_OK;
}

static int finish_stage(sdbv_ctx *ctx, Table *t) {
	uint64_t nb = (t->n + THREADS - 1) / THREADS;
	if (t->metric == SDBV_METRIC_COSINE) {
		hipLaunchKernelGGL(k_norms, dim3((uint32_t)nb), dim3(THREADS), 0,
		                   ctx->stream, t->cm, t->norms, t->n, t->n_pad, t->d);
	}
	if (t->metric <= SDBV_METRIC_EUCLIDEAN) {
		// batch-path selection aux (cosine 1/norm, euclidean sumsq); the
		// other metrics take the all-distances route and need none
		HIP_CHECK(ctx, hipMalloc(&t->aux, t->n_pad * sizeof(float)));
		t->bytes += t->n_pad * sizeof(float);
		hipLaunchKernelGGL(k_aux, dim3((uint32_t)nb), dim3(THREADS), 0,
		                   ctx->stream, t->cm, t->norms, t->aux, t->n,
		                   t->n_pad, t->d, (int)t->metric);
	}
	HIP_CHECK(ctx, hipStreamSynchronize(ctx->stream));
	HIP_CHECK(ctx, hipGetLastError());
	uint64_t total = 0;
	for (auto &kv : ctx->tables)
		total += kv.second.bytes;
	ctx->stats.bytes_staged = total;
	return SDBV_OK;
}

int sdbv_stage_corpus(sdbv_ctx *ctx, uint64_t table, const float *rows,
                      const uint64_t *ids, uint64_t n, uint32_t d,
                      uint8_t metric) {
	std::lock_guard<std::mutex> lk(ctx->mu);
	// tie-break contract: ids must be strictly increasing (see include/sdbv.h)
	if (ids)
		for (uint64_t i = 1; i < n; i++)
			if (ids[i] <= ids[i - 1])
				return SDBV_ERR_IDS_UNSORTED;
	Table *t = nullptr;
	int rc = stage_common(ctx, table, n, d, metric, &t);
	if (rc != SDBV_OK)
		return rc;
	// upload row-major then transpose on device
	float *rm = nullptr;
	HIP_CHECK(ctx, hipMalloc(&rm, n * (uint64_t)d * sizeof(float)));
	HIP_CHECK(ctx, hipMemcpyAsync(rm, rows, n * (uint64_t)d * sizeof(float),
	                              hipMemcpyHostToDevice, ctx->stream));
	dim3 grid((uint32_t)((t->n_pad + 31) / 32), (d + 31) / 32);
	hipLaunchKernelGGL(k_transpose, grid, dim3(256), 0, ctx->stream, rm, t->cm,
	                   n, t->n_pad, d);
	if (ids) {
		HIP_CHECK(ctx, hipMemcpyAsync(t->ids_dev, ids, n * sizeof(uint64_t),
		                              hipMemcpyHostToDevice, ctx->stream));
	} else {
		uint64_t nb = (t->n_pad + THREADS - 1) / THREADS;
		hipLaunchKernelGGL(k_fill_ids, dim3((uint32_t)nb), dim3(THREADS), 0,
		                   ctx->stream, t->ids_dev, t->n_pad, 0);
	}
	int rc2 = finish_stage(ctx, t);
	(void)hipFree(rm);
	return rc2;
}

int sdbv_stage_synthetic(sdbv_ctx *ctx, uint64_t table, uint64_t n, uint32_t d,
                         uint8_t metric, uint64_t seed, uint64_t row_offset,
                         uint64_t id_base) {
	std::lock_guard<std::mutex> lk(ctx->mu);
	Table *t = nullptr;
	int rc = stage_common(ctx, table, n, d, metric, &t);
	if (rc != SDBV_OK)
		return rc;
	hipLaunchKernelGGL(k_gen_cm, dim3(8192), dim3(256), 0, ctx->stream, t->cm,
	                   t->n, t->n_pad, t->d, seed, row_offset);
	uint64_t nb = (t->n_pad + THREADS - 1) / THREADS;
	hipLaunchKernelGGL(k_fill_ids, dim3((uint32_t)nb), dim3(THREADS), 0,
	                   ctx->stream, t->ids_dev, t->n_pad, id_base);
	return finish_stage(ctx, t);
}

uint64_t sdbv_table_rows(sdbv_ctx *ctx, uint64_t table) {
	std::lock_guard<std::mutex> lk(ctx->mu);
	auto it = ctx->tables.find(table);
	return it == ctx->tables.end() ? 0 : it->second.n;
}

int sdbv_drop_table(sdbv_ctx *ctx, uint64_t table) {
	std::lock_guard<std::mutex> lk(ctx->mu);
	auto it = ctx->tables.find(table);
	if (it == ctx->tables.end())
		return SDBV_ERR_NO_TABLE;
	free_table(it->second);
	ctx->tables.erase(it);
	return SDBV_OK;
}

int sdbv_table_set_order(sdbv_ctx *ctx, uint64_t table, double order) {
	std::lock_guard<std::mutex> lk(ctx->mu);
	auto it = ctx->tables.find(table);
	if (it == ctx->tables.end())
		return SDBV_ERR_NO_TABLE;
	it->second.order = order;
	return SDBV_OK;
}

static int ensure_query_scratch(sdbv_ctx *ctx, uint32_t d, uint64_t nblocks,
                                uint32_t k) {
	if (ctx->q_cap < d) {
		if (ctx->q_dev)
			(void)hipFree(ctx->q_dev);
		HIP_CHECK(ctx, hipMalloc(&ctx->q_dev, d * sizeof(float)));
		if (ctx->q_pin)
			(void)hipHostFree(ctx->q_pin);
		HIP_CHECK(ctx, hipHostMalloc(&ctx->q_pin, d * sizeof(float)));
		ctx->q_cap = d;
	}
	uint64_t need = nblocks * k * sizeof(Cand);
	if (ctx->block_out_cap < need) {
		if (ctx->block_out)
			(void)hipFree(ctx->block_out);
		HIP_CHECK(ctx, hipMalloc(&ctx->block_out, need));
		ctx->block_out_cap = need;
	}
	if (!ctx->final_out)
		HIP_CHECK(ctx, hipMalloc(&ctx->final_out, MAX_K * sizeof(Cand)));
	if (!ctx->merge_tmp)
		HIP_CHECK(ctx,
		          hipMalloc(&ctx->merge_tmp, 32 * MAX_K * sizeof(Cand)));
	return SDBV_OK;
}

// Launch the all-distances computation for any metric (q already staged in
// ctx->q_dev). Jaccard runs its block-per-row LDS-set kernel; the rest run
// the per-lane k_all_dists chain.
static void launch_all_dists(sdbv_ctx *ctx, Table &t, const float *q,
                             double *dout) {
	uint64_t nb = (t.n + THREADS - 1) / THREADS;
	double q_norm = t.metric == SDBV_METRIC_COSINE
	                    ? sqrt((double)h_sumsq_f32(q, t.d))
	                    : 0;
	double q_mean = t.metric == SDBV_METRIC_PEARSON ? h_mean_f32(q, t.d) : 0;
#define LAUNCH_AD(M)                                                       \
	hipLaunchKernelGGL(k_all_dists<M>, dim3((uint32_t)nb), dim3(THREADS), \
	                   0, ctx->stream, t.cm, t.norms, t.n, t.n_pad, t.d,  \
	                   ctx->q_dev, q_norm, t.order, q_mean, dout)
	switch (t.metric) {
	case SDBV_METRIC_COSINE: LAUNCH_AD(0); break;
	case SDBV_METRIC_EUCLIDEAN: LAUNCH_AD(1); break;
	case SDBV_METRIC_MANHATTAN: LAUNCH_AD(2); break;
	case SDBV_METRIC_CHEBYSHEV: LAUNCH_AD(3); break;
	case SDBV_METRIC_HAMMING: LAUNCH_AD(4); break;
	case SDBV_METRIC_JACCARD:
		hipLaunchKernelGGL(k_jaccard_dists, dim3((uint32_t)t.n), dim3(256),
		                   0, ctx->stream, t.cm, t.n, t.n_pad, t.d,
		                   ctx->q_dev, dout);
		break;
	case SDBV_METRIC_MINKOWSKI: LAUNCH_AD(6); break;
	case SDBV_METRIC_PEARSON: LAUNCH_AD(7); break;
	}
#undef LAUNCH_AD
}

// All-distances launch + exact host selection: the route for k beyond the
// scan kernel's LDS top-K window (MAX_K), and the product path proper for
// the six non-headline metrics (SURVEY §8 a2 closure). ids are strictly
// increasing, so (dist, row) order equals the contract's (dist, id)
// order. Costs one n-f64 D2H per query. Caller holds ctx->mu.
static int knn_large_k(sdbv_ctx *ctx, Table &t, const float *q, uint32_t d,
                       uint32_t k, uint64_t *out_ids, double *out_dists,
                       uint32_t *out_n) {
	int rc = ensure_query_scratch(ctx, d, 1, 1);
	if (rc != SDBV_OK)
		return rc;
	std::memcpy(ctx->q_pin, q, d * sizeof(float)); // pinned staging
	HIP_CHECK(ctx, hipMemcpyAsync(ctx->q_dev, ctx->q_pin, d * sizeof(float),
	                              hipMemcpyHostToDevice, ctx->stream));
	double *dout = nullptr;
	HIP_CHECK(ctx, hipMalloc(&dout, t.n * sizeof(double)));
	HIP_CHECK(ctx, hipEventRecord(ctx->ev0, ctx->stream));
	launch_all_dists(ctx, t, q, dout);
	HIP_CHECK(ctx, hipEventRecord(ctx->ev1, ctx->stream));
	std::vector<double> hd(t.n);
	std::vector<uint64_t> hids(t.n);
	if (hipMemcpyAsync(hd.data(), dout, t.n * sizeof(double),
	                   hipMemcpyDeviceToHost, ctx->stream) != hipSuccess ||
	    hipMemcpyAsync(hids.data(), t.ids_dev, t.n * sizeof(uint64_t),
	                   hipMemcpyDeviceToHost, ctx->stream) != hipSuccess ||
	    hipStreamSynchronize(ctx->stream) != hipSuccess ||
	    hipGetLastError() != hipSuccess) {
		(void)hipFree(dout);
		ctx->err = "knn_large_k: device error";
		return SDBV_ERR_HIP;
	}
	(void)hipFree(dout);
	float ms = 0;
	(void)hipEventElapsedTime(&ms, ctx->ev0, ctx->ev1);
	ctx->stats.last_scan_kernel_ms = ms;
	ctx->stats.last_merge_kernel_ms = 0;
	ctx->stats.last_rows_scanned = t.n;
	// bounded max-heap of (total_key(dist), row): keep the k smallest
	auto tkey = [](double x) {
		uint64_t bits;
		std::memcpy(&bits, &x, 8);
		return (bits >> 63) ? ~bits : (bits | 0x8000000000000000ULL);
	};
	std::priority_queue<std::pair<uint64_t, uint64_t>> heap;
	for (uint64_t r = 0; r < t.n; r++) {
		std::pair<uint64_t, uint64_t> e{tkey(hd[r]), r};
		if (heap.size() < k) {
			heap.push(e);
		} else if (e < heap.top()) {
			heap.pop();
			heap.push(e);
		}
	}
	uint32_t m = (uint32_t)heap.size();
	*out_n = m;
	for (uint32_t i = m; i-- > 0;) {
		auto e = heap.top();
		heap.pop();
		out_ids[i] = hids[e.second];
		out_dists[i] = hd[e.second];
	}
	return SDBV_OK;
}

int sdbv_knn_bruteforce(sdbv_ctx *ctx, uint64_t table, const float *q,
                        uint32_t d, uint32_t k, uint64_t *out_ids,
                        double *out_dists, uint32_t *out_n) {
	std::lock_guard<std::mutex> lk(ctx->mu);
	auto it = ctx->tables.find(table);
	if (it == ctx->tables.end())
		return SDBV_ERR_NO_TABLE;
	Table &t = it->second;
	if (d != t.d)
		return SDBV_ERR_BAD_ARG;
	if (k == 0 || k > (1u << 22))
		return SDBV_ERR_BAD_ARG;
	if (k > MAX_K || t.metric > SDBV_METRIC_EUCLIDEAN)
		return knn_large_k(ctx, t, q, d, k, out_ids, out_dists, out_n);

	// pick a grid: >=2048 blocks to fill 256 CUs, contiguous spans per block
	uint64_t tiles = (t.n + TILE - 1) / TILE;
	uint64_t nblocks = tiles < 2048 ? tiles : 2048;
	if (nblocks == 0)
		nblocks = 1;
	uint64_t tiles_per_block = (tiles + nblocks - 1) / nblocks;
	uint64_t rows_per_block = tiles_per_block * TILE;
	nblocks = (t.n + rows_per_block - 1) / rows_per_block;

	int rc = ensure_query_scratch(ctx, d, nblocks, k);
	if (rc != SDBV_OK)
		return rc;
	std::memcpy(ctx->q_pin, q, d * sizeof(float)); // pinned staging
	HIP_CHECK(ctx, hipMemcpyAsync(ctx->q_dev, ctx->q_pin, d * sizeof(float),
	                              hipMemcpyHostToDevice, ctx->stream));
	double q_norm = sqrt((double)h_sumsq_f32(q, d));

	auto t_start = std::chrono::steady_clock::now();
	HIP_CHECK(ctx, hipEventRecord(ctx->ev0, ctx->stream));
	if (t.metric == SDBV_METRIC_COSINE)
		hipLaunchKernelGGL(k_scan<0>, dim3((uint32_t)nblocks), dim3(THREADS), 0,
		                   ctx->stream, t.cm, t.norms, t.n, t.n_pad, t.d,
		                   ctx->q_dev, q_norm, k, rows_per_block,
		                   (Cand *)ctx->block_out, t.ids_dev);
	else
		hipLaunchKernelGGL(k_scan<1>, dim3((uint32_t)nblocks), dim3(THREADS), 0,
		                   ctx->stream, t.cm, t.norms, t.n, t.n_pad, t.d,
		                   ctx->q_dev, q_norm, k, rows_per_block,
		                   (Cand *)ctx->block_out, t.ids_dev);
	HIP_CHECK(ctx, hipEventRecord(ctx->ev1, ctx->stream));
	{
		uint64_t total = nblocks * k;
		const uint64_t G = 32; // level-1 fan-in
		if (total > 4096 && nblocks > G) {
			// two-level tree merge: G slices in parallel, then one block
			// over G*k (scratch lives past final_out's k entries)
			uint64_t per = ((nblocks + G - 1) / G) * k;
			hipLaunchKernelGGL(k_merge, dim3((uint32_t)G), dim3(THREADS),
			                   0, ctx->stream, (Cand *)ctx->block_out,
			                   total, k, per, (Cand *)ctx->merge_tmp);
			hipLaunchKernelGGL(k_merge, dim3(1), dim3(THREADS), 0,
			                   ctx->stream, (Cand *)ctx->merge_tmp, G * k,
			                   k, G * k, (Cand *)ctx->final_out);
		} else {
			hipLaunchKernelGGL(k_merge, dim3(1), dim3(THREADS), 0,
			                   ctx->stream, (Cand *)ctx->block_out, total,
			                   k, total, (Cand *)ctx->final_out);
		}
	}
	HIP_CHECK(ctx, hipEventRecord(ctx->ev2, ctx->stream));
	Cand host_out[MAX_K];
	HIP_CHECK(ctx, hipMemcpyAsync(host_out, ctx->final_out, k * sizeof(Cand),
	                              hipMemcpyDeviceToHost, ctx->stream));
	HIP_CHECK(ctx, hipStreamSynchronize(ctx->stream));
	HIP_CHECK(ctx, hipGetLastError());

	float ms_scan = 0, ms_merge = 0;
	(void)hipEventElapsedTime(&ms_scan, ctx->ev0, ctx->ev1);
	(void)hipEventElapsedTime(&ms_merge, ctx->ev1, ctx->ev2);
	ctx->stats.last_scan_kernel_ms = ms_scan;
	ctx->stats.last_merge_kernel_ms = ms_merge;
	ctx->stats.last_rows_scanned = t.n;
	ctx->stats.last_total_ms =
	    std::chrono::duration<double, std::milli>(
	        std::chrono::steady_clock::now() - t_start)
	        .count();

	uint32_t m = (uint32_t)(k < t.n ? k : t.n);
	*out_n = m;
	for (uint32_t i = 0; i < m; i++) {
		out_ids[i] = host_out[i].id;
		out_dists[i] = host_out[i].dist;
	}
	return SDBV_OK;
}

// Debug/parity: dump all n distances (small n only; n doubles to host).
int sdbv_all_distances(sdbv_ctx *ctx, uint64_t table, const float *q,
                       uint32_t d, double *out) {
	std::lock_guard<std::mutex> lk(ctx->mu);
	auto it = ctx->tables.find(table);
	if (it == ctx->tables.end())
		return SDBV_ERR_NO_TABLE;
	Table &t = it->second;
	if (d != t.d)
		return SDBV_ERR_BAD_ARG;
	int rc = ensure_query_scratch(ctx, d, 1, 1);
	if (rc != SDBV_OK)
		return rc;
	std::memcpy(ctx->q_pin, q, d * sizeof(float)); // pinned staging
	HIP_CHECK(ctx, hipMemcpyAsync(ctx->q_dev, ctx->q_pin, d * sizeof(float),
	                              hipMemcpyHostToDevice, ctx->stream));
	double *dout = nullptr;
	HIP_CHECK(ctx, hipMalloc(&dout, t.n * sizeof(double)));
	launch_all_dists(ctx, t, q, dout);
	HIP_CHECK(ctx, hipMemcpyAsync(out, dout, t.n * sizeof(double),
	                              hipMemcpyDeviceToHost, ctx->stream));
	HIP_CHECK(ctx, hipStreamSynchronize(ctx->stream));
	HIP_CHECK(ctx, hipGetLastError());
	(void)hipFree(dout);
	return SDBV_OK;
}

int sdbv_gather_distance(sdbv_ctx *ctx, uint64_t table, const uint32_t *rows,
                         uint32_t nrows, const float *q, uint32_t d,
                         double *out_dists) {
	std::lock_guard<std::mutex> lk(ctx->mu);
	auto it = ctx->tables.find(table);
	if (it == ctx->tables.end())
		return SDBV_ERR_NO_TABLE;
	Table &t = it->second;
	if (d != t.d || nrows == 0)
		return SDBV_ERR_BAD_ARG;
	int rc = ensure_query_scratch(ctx, d, 1, 1);
	if (rc != SDBV_OK)
		return rc;
	std::memcpy(ctx->q_pin, q, d * sizeof(float)); // pinned staging
	HIP_CHECK(ctx, hipMemcpyAsync(ctx->q_dev, ctx->q_pin, d * sizeof(float),
	                              hipMemcpyHostToDevice, ctx->stream));
	double q_norm = sqrt((double)h_sumsq_f32(q, d));
	uint32_t *rows_dev = nullptr;
	double *dout = nullptr;
	HIP_CHECK(ctx, hipMalloc(&rows_dev, nrows * sizeof(uint32_t)));
	HIP_CHECK(ctx, hipMalloc(&dout, nrows * sizeof(double)));
	HIP_CHECK(ctx, hipMemcpyAsync(rows_dev, rows, nrows * sizeof(uint32_t),
	                              hipMemcpyHostToDevice, ctx->stream));
	if (t.metric == SDBV_METRIC_COSINE)
		hipLaunchKernelGGL(k_gather_dist<0>, dim3(nrows), dim3(64), 0,
		                   ctx->stream, t.cm, t.norms, t.n_pad, t.d, rows_dev,
		                   nrows, ctx->q_dev, q_norm, dout);
	else
		hipLaunchKernelGGL(k_gather_dist<1>, dim3(nrows), dim3(64), 0,
		                   ctx->stream, t.cm, t.norms, t.n_pad, t.d, rows_dev,
		                   nrows, ctx->q_dev, q_norm, dout);
	HIP_CHECK(ctx, hipMemcpyAsync(out_dists, dout, nrows * sizeof(double),
	                              hipMemcpyDeviceToHost, ctx->stream));
	HIP_CHECK(ctx, hipStreamSynchronize(ctx->stream));
	HIP_CHECK(ctx, hipGetLastError());
	(void)hipFree(rows_dev);
	(void)hipFree(dout);
	return SDBV_OK;
}

static int ensure_cap(sdbv_ctx *ctx, void **buf, uint64_t *cap, uint64_t need) {
	if (*cap >= need)
		return SDBV_OK;
	if (*buf)
		(void)hipFree(*buf);
	*buf = nullptr;
	*cap = 0;
	HIP_CHECK(ctx, hipMalloc(buf, need));
	*cap = need;
	return SDBV_OK;
}

int sdbv_knn_batch(sdbv_ctx *ctx, uint64_t table, const float *Q, uint32_t b,
                   uint32_t d, uint32_t k, uint64_t *out_ids,
                   double *out_dists) {
	std::lock_guard<std::mutex> lk(ctx->mu);
	auto it = ctx->tables.find(table);
	if (it == ctx->tables.end())
		return SDBV_ERR_NO_TABLE;
	Table &t = it->second;
	if (d != t.d || b == 0 || k == 0 || k > MAX_K - BATCH_SLACK)
		return SDBV_ERR_BAD_ARG;
	if (t.metric > SDBV_METRIC_EUCLIDEAN) {
		// non-GEMM metrics: per-query all-distances route (exact; the MFMA
		// batch path is only for the genuinely dense cosine/euclidean case)
		for (uint32_t j = 0; j < b; j++) {
			uint32_t m = 0;
			int rc = knn_large_k(ctx, t, Q + (uint64_t)j * d, d, k,
			                     out_ids + (uint64_t)j * k,
			                     out_dists + (uint64_t)j * k, &m);
			if (rc)
				return rc;
			for (uint32_t i = m; i < k; i++) {
				out_ids[(uint64_t)j * k + i] = ~0ULL;
				out_dists[(uint64_t)j * k + i] =
				    std::numeric_limits<double>::infinity();
			}
		}
		return SDBV_OK;
	}
	const int kk = (int)k + BATCH_SLACK;

	if (!ctx->blas) {
		if (rocblas_create_handle(&ctx->blas) != rocblas_status_success) {
			ctx->err = "rocblas_create_handle failed";
			return SDBV_ERR_HIP;
		}
		rocblas_set_stream(ctx->blas, ctx->stream);
	}

	// chunk size: multiple of TILE, bounded scratch (<=1 GiB at b=1024)
	uint64_t chunk = 262144;
	if (chunk > t.n_pad)
		chunk = t.n_pad;
	int rc;
	if ((rc = ensure_cap(ctx, (void **)&ctx->S, &ctx->S_cap,
	                     chunk * b * sizeof(float))))
		return rc;
	if ((rc = ensure_cap(ctx, (void **)&ctx->bstate, &ctx->bstate_cap,
	                     (uint64_t)b * kk * sizeof(BCand))))
		return rc;
	if ((rc = ensure_cap(ctx, (void **)&ctx->Q_dev, &ctx->Q_cap,
	                     (uint64_t)b * d * sizeof(float))))
		return rc;
	if ((rc = ensure_cap(ctx, (void **)&ctx->qnorms, &ctx->qnorms_cap,
	                     (uint64_t)b * sizeof(double))))
		return rc;
	if ((rc = ensure_cap(ctx, (void **)&ctx->bdists, &ctx->bdists_cap,
	                     (uint64_t)b * kk * sizeof(double))))
		return rc;

	HIP_CHECK(ctx, hipMemcpyAsync(ctx->Q_dev, Q,
	                              (uint64_t)b * d * sizeof(float),
	                              hipMemcpyHostToDevice, ctx->stream));
	hipLaunchKernelGGL(k_qnorms, dim3((b + 255) / 256), dim3(256), 0,
	                   ctx->stream, ctx->Q_dev, b, d, ctx->qnorms);
	{
		uint64_t total = (uint64_t)b * kk;
		hipLaunchKernelGGL(k_binit, dim3((uint32_t)((total + 255) / 256)),
		                   dim3(256), 0, ctx->stream, (BCand *)ctx->bstate,
		                   total);
	}

	// Fused MFMA path (default): bootstrap per-query thresholds on the
	// first rows via the legacy rocBLAS + k_batch_topk pass, then ONE
	// hand-written MFMA launch (k_mfma_scan_topk) covers the rest — the
	// score matrix never touches HBM. SDBV_BATCH_LEGACY=1 forces the
	// round-1 path (A/B comparisons); shape constraints fall back too.
	const bool use_fused = (d % FMM_BK) == 0 && (b % FMM_BN) == 0 &&
	                       !std::getenv("SDBV_BATCH_LEGACY");
	uint64_t fused_row0 = t.n; // rows from here on go to the fused kernel
	if (use_fused && t.n > 65536)
		fused_row0 = 65536; // multiple of FMM_BM and TILE

	double ms_gemm = 0, ms_select = 0;
	auto legacy_rows = [&](uint64_t from, uint64_t to) -> int {
		// `to` == t.n means "to the end": extend the last chunk over the
		// padded rows exactly like the round-1 loop (k_batch_topk's n
		// guard skips them); a bounded `to` (the TILE-aligned bootstrap)
		// is covered exactly.
		for (uint64_t row0 = from; row0 < to; row0 += chunk) {
			uint64_t lim = (to == t.n) ? t.n_pad : to;
			uint32_t nc = (uint32_t)std::min<uint64_t>(chunk, lim - row0);
			const float alpha = 1.0f, beta = 0.0f;
			HIP_CHECK(ctx, hipEventRecord(ctx->ev0, ctx->stream));
			// S[nc x b] = corpus[nc x d] (col-major view of cm, lda n_pad)
			//           x Q^T[d x b]    (col-major view of row-major Q)
			if (rocblas_sgemm(ctx->blas, rocblas_operation_none,
			                  rocblas_operation_none, (rocblas_int)nc,
			                  (rocblas_int)b, (rocblas_int)d, &alpha,
			                  t.cm + row0, (rocblas_int)t.n_pad, ctx->Q_dev,
			                  (rocblas_int)d, &beta, ctx->S,
			                  (rocblas_int)nc) != rocblas_status_success) {
				ctx->err = "rocblas_sgemm failed";
				return SDBV_ERR_HIP;
			}
			HIP_CHECK(ctx, hipEventRecord(ctx->ev1, ctx->stream));
			hipLaunchKernelGGL(k_batch_topk, dim3(b), dim3(THREADS), 0,
			                   ctx->stream, ctx->S, t.aux, row0, nc, t.n,
			                   (int)t.metric, (BCand *)ctx->bstate, kk);
			HIP_CHECK(ctx, hipEventRecord(ctx->ev2, ctx->stream));
			HIP_CHECK(ctx, hipStreamSynchronize(ctx->stream));
			float m0 = 0, m1 = 0;
			(void)hipEventElapsedTime(&m0, ctx->ev0, ctx->ev1);
			(void)hipEventElapsedTime(&m1, ctx->ev1, ctx->ev2);
			ms_gemm += m0;
			ms_select += m1;
		}
		return SDBV_OK;
	};
	if ((rc = legacy_rows(0, fused_row0)))
		return rc;
	if (fused_row0 < t.n) {
		if ((rc = ensure_cap(ctx, (void **)&ctx->btheta, &ctx->btheta_cap,
		                     (uint64_t)b * sizeof(uint32_t))))
			return rc;
		if ((rc = ensure_cap(ctx, (void **)&ctx->bcand, &ctx->bcand_cap,
		                     (uint64_t)b * FMM_CAND_CAP * sizeof(BCand))))
			return rc;
		if ((rc = ensure_cap(ctx, (void **)&ctx->bcand_cnt,
		                     &ctx->bcand_cnt_cap,
		                     (uint64_t)b * sizeof(uint32_t))))
			return rc;
		hipLaunchKernelGGL(k_theta_init, dim3((b + 255) / 256), dim3(256),
		                   0, ctx->stream, (const BCand *)ctx->bstate, kk, b,
		                   ctx->btheta);
		HIP_CHECK(ctx, hipMemsetAsync(ctx->bcand_cnt, 0,
		                              b * sizeof(uint32_t), ctx->stream));
		uint32_t gx = (uint32_t)((t.n - fused_row0 + FMM_BM - 1) / FMM_BM);
		HIP_CHECK(ctx, hipEventRecord(ctx->ev0, ctx->stream));
		hipLaunchKernelGGL(k_mfma_scan_topk, dim3(gx, b / FMM_BN),
		                   dim3(256), 0, ctx->stream, t.cm, t.n_pad, d,
		                   ctx->Q_dev, b, fused_row0, t.n, t.aux,
		                   (int)t.metric, ctx->btheta, (BCand *)ctx->bcand,
		                   ctx->bcand_cnt);
		HIP_CHECK(ctx, hipEventRecord(ctx->ev1, ctx->stream));
		hipLaunchKernelGGL(k_cand_fold, dim3(b), dim3(THREADS), 0,
		                   ctx->stream, (const BCand *)ctx->bcand,
		                   ctx->bcand_cnt, (BCand *)ctx->bstate, kk);
		HIP_CHECK(ctx, hipEventRecord(ctx->ev2, ctx->stream));
		std::vector<uint32_t> h_cnt(b);
		HIP_CHECK(ctx, hipMemcpyAsync(h_cnt.data(), ctx->bcand_cnt,
		                              b * sizeof(uint32_t),
		                              hipMemcpyDeviceToHost, ctx->stream));
		HIP_CHECK(ctx, hipStreamSynchronize(ctx->stream));
		HIP_CHECK(ctx, hipGetLastError());
		float m0 = 0, m1 = 0;
		(void)hipEventElapsedTime(&m0, ctx->ev0, ctx->ev1);
		(void)hipEventElapsedTime(&m1, ctx->ev1, ctx->ev2);
		ms_gemm += m0;
		ms_select += m1;
		bool overflow = false;
		for (uint32_t j = 0; j < b; j++)
			if (h_cnt[j] > FMM_CAND_CAP)
				overflow = true;
		if (overflow) {
			// a candidate buffer filled (adversarially ordered corpus):
			// re-present the fused rows through the legacy pass — state
			// merging is exact, so dropped appends are recovered
			if ((rc = legacy_rows(fused_row0, t.n)))
				return rc;
		}
	}

	// exact recompute of survivors with the restated chain
	HIP_CHECK(ctx, hipEventRecord(ctx->ev0, ctx->stream));
	hipLaunchKernelGGL(k_batch_exact, dim3(b, kk), dim3(64), 0, ctx->stream,
	                   t.cm, t.norms, t.n_pad, t.d, ctx->Q_dev, ctx->qnorms,
	                   (int)t.metric, (const BCand *)ctx->bstate, kk,
	                   ctx->bdists);
	HIP_CHECK(ctx, hipEventRecord(ctx->ev1, ctx->stream));

	std::vector<BCand> h_state((uint64_t)b * kk);
	std::vector<double> h_dists((uint64_t)b * kk);
	HIP_CHECK(ctx, hipMemcpyAsync(h_state.data(), ctx->bstate,
	                              h_state.size() * sizeof(BCand),
	                              hipMemcpyDeviceToHost, ctx->stream));
	HIP_CHECK(ctx, hipMemcpyAsync(h_dists.data(), ctx->bdists,
	                              h_dists.size() * sizeof(double),
	                              hipMemcpyDeviceToHost, ctx->stream));
	HIP_CHECK(ctx, hipStreamSynchronize(ctx->stream));
	HIP_CHECK(ctx, hipGetLastError());
	float me = 0;
	(void)hipEventElapsedTime(&me, ctx->ev0, ctx->ev1);
	ctx->ms_gemm = ms_gemm;
	ctx->ms_select = ms_select;
	ctx->ms_exact = me;
	ctx->stats.last_scan_kernel_ms = ms_gemm; // dominant kernel of the batch
	ctx->stats.last_merge_kernel_ms = ms_select;
	ctx->stats.last_rows_scanned = t.n;

	// final exact rank per query: ascending (total_cmp(dist), id)
	std::vector<uint64_t> ids_host(t.n_pad);
	HIP_CHECK(ctx, hipMemcpy(ids_host.data(), t.ids_dev,
	                         t.n_pad * sizeof(uint64_t),
	                         hipMemcpyDeviceToHost));
	auto total_key = [](double x) {
		uint64_t bits;
		std::memcpy(&bits, &x, 8);
		return (bits >> 63) ? ~bits : (bits | 0x8000000000000000ULL);
	};
	std::vector<std::pair<std::pair<uint64_t, uint64_t>, double>> cand(kk);
	for (uint32_t j = 0; j < b; j++) {
		size_t m = 0;
		for (int c = 0; c < kk; c++) {
			BCand bc = h_state[(uint64_t)j * kk + c];
			if (bc.row == ~0u)
				continue;
			double dd = h_dists[(uint64_t)j * kk + c];
			cand[m++] = {{total_key(dd), ids_host[bc.row]}, dd};
		}
		std::sort(cand.begin(), cand.begin() + m);
		uint32_t out_m = (uint32_t)std::min<size_t>(k, m);
		for (uint32_t c = 0; c < out_m; c++) {
			out_ids[(uint64_t)j * k + c] = cand[c].first.second;
			out_dists[(uint64_t)j * k + c] = cand[c].second;
		}
		for (uint32_t c = out_m; c < k; c++) {
			out_ids[(uint64_t)j * k + c] = ~0ULL;
			out_dists[(uint64_t)j * k + c] =
			    std::numeric_limits<double>::infinity();
		}
	}
	return SDBV_OK;
}
} // extern "C"

// ===========================================================================
// HNSW — product implementation of the reference graph algorithm
// (hnsw/mod.rs, layer.rs, heuristic.rs, knn.rs queues). Host-side topology +
// build (as in the reference engine); layer-0 query expansion = the GPU
// batched gather+distance kernel. Independent of oracle/ (the oracle is the
// parity checker for this code, never a dependency).
// ===========================================================================

namespace hnsw {

// f64 total_cmp key (knn.rs:128-160)
static inline uint64_t total_key(double x) {
	uint64_t bits;
	std::memcpy(&bits, &x, 8);
	return (bits >> 63) ? ~bits : (bits | 0x8000000000000000ULL);
}

// DoublePriorityQueue restatement (knn.rs:15-123): ordered by total_cmp(dist),
// FIFO within equal distance (push order); pop_last removes the LATEST of
// the max key. Implemented as a sorted (key, seq) vector with a lazy head —
// exactly the BTreeMap<FloatKey, VecDeque> observable order, without the
// per-node allocations (this queue is the host build/search hot path).
struct PQ {
	struct E {
		uint64_t key;
		uint32_t seq;
		uint32_t id;
		double d;
	};
	std::vector<E> v; // ascending (key, seq); v[head..] is the live queue
	size_t head = 0;
	uint32_t next_seq = 0;
	size_t n = 0;
	void push(double d, uint32_t id) {
		E e{total_key(d), next_seq++, id, d};
		auto it = std::upper_bound(
		    v.begin() + head, v.end(), e, [](const E &a, const E &b) {
			    return a.key != b.key ? a.key < b.key : a.seq < b.seq;
		    });
		v.insert(it, e);
		n++;
	}
	bool pop_first(double *d, uint32_t *id) {
		if (n == 0)
			return false;
		*d = v[head].d;
		*id = v[head].id;
		head++;
		n--;
		return true;
	}
	void pop_last() { // latest push of the max key == max (key, seq)
		if (n == 0)
			return;
		v.pop_back();
		n--;
	}
	bool peek_first(double *d, uint32_t *id) const {
		if (n == 0)
			return false;
		*d = v[head].d;
		*id = v[head].id;
		return true;
	}
	double peek_last_dist(double fb) const {
		return n == 0 ? fb : v.back().d;
	}
	std::vector<std::pair<double, uint32_t>> to_vec() const {
		std::vector<std::pair<double, uint32_t>> out;
		out.reserve(n);
		for (size_t i = head; i < v.size(); i++)
			out.push_back({v[i].d, v[i].id});
		return out;
	}
};

// Restated per-row distance chain on the host (same ops as the device
// kernels; the whole translation unit is compiled -ffp-contract=off).
static double host_dot_f32(const float *a, const float *b, uint32_t d) {
	float p[8] = {0, 0, 0, 0, 0, 0, 0, 0};
	uint32_t i = 0;
	for (; i + 8 <= d; i += 8)
		for (uint32_t t = 0; t < 8; t++)
			p[t] += a[i + t] * b[i + t];
	float s = 0;
	s += (p[0] + p[4]);
	s += (p[1] + p[5]);
	s += (p[2] + p[6]);
	s += (p[3] + p[7]);
	for (; i < d; i++)
		s += a[i] * b[i];
	return (double)s;
}
static double host_sumsq_f32(const float *a, uint32_t d) {
	float p[8] = {0, 0, 0, 0, 0, 0, 0, 0};
	uint32_t i = 0;
	for (; i + 8 <= d; i += 8)
		for (uint32_t t = 0; t < 8; t++)
			p[t] += a[i + t] * a[i + t];
	float s = 0;
	s += ((p[0] + p[4]) + (p[1] + p[5]));
	s += ((p[2] + p[6]) + (p[3] + p[7]));
	for (; i < d; i++)
		s += a[i] * a[i];
	return (double)s;
}

struct Layer {
	std::vector<std::vector<uint32_t>> edges;
	uint32_t m_max;
	// graph.rs nodes-map membership: get_edges is None for absent nodes and
	// layer.remove is a no-op on them. Kept alongside `edges` (which is
	// sized to the element count for lock-striped parallel inserts).
	std::vector<uint8_t> in_layer;
	bool has(uint32_t id) const {
		return id < in_layer.size() && in_layer[id];
	}
};

} // namespace hnsw

struct sdbv_hnsw {
	sdbv_ctx *ctx;
	uint32_t d;
	uint8_t metric;
	uint32_t m, m0, efc;
	bool extend, keep;
	double ml;
	uint64_t rng_state;
	std::vector<float> vecs;    // host row-major copy (build + upper layers)
	std::vector<double> norms;  // per-element f64 norm (cosine)
	std::vector<hnsw::Layer> layers;
	// elements-map membership (hnsw/elements.rs): remove() keeps the vector
	// slot but the element no longer exists for searches
	std::vector<uint8_t> elem_present;
	bool dirty = false; // host graph changed since finalize (device stale)
	int64_t enter_point = -1;
	uint64_t next_id = 0;
	// parallel build
	std::vector<std::mutex> node_locks{4096};
	std::mutex global_mu;
	// device side (after finalize)
	uint64_t table = ~0ULL;
	bool finalized = false;
	uint32_t *rows_dev = nullptr;
	double *dout_dev = nullptr;
	float *q_dev = nullptr;
	uint32_t *rows_pinned = nullptr; // pinned host staging (per-hop latency)
	double *dists_pinned = nullptr;
	// persistent-kernel search state (device graph + row-major vectors)
	float *rm_dev = nullptr;        // [n][d] row-major
	uint32_t *offsets_dev = nullptr; // layer-0 CSR
	uint32_t *edges_dev = nullptr;
	double *norms_dev = nullptr;    // per-element f64 norm (cosine)
	uint32_t *vis_dev = nullptr;    // visited bitsets scratch
	uint64_t vis_cap = 0;
	uint32_t max_deg = 0; // actual max layer-0 degree at finalize (scratch
	                      // sizing; a parallel build can transiently exceed
	                      // m0 — see layer_insert_apply's keep-back)
	// GPU snapshot-build state (padded layer-0 adjacency, delta-updated)
	uint32_t *adj_dev = nullptr; // [n][adj_stride]
	uint32_t *deg_dev = nullptr; // [n]
	uint32_t adj_stride = 0;
	uint64_t adj_nodes = 0; // allocated node capacity of adj_dev/deg_dev
	uint64_t dev_rows = 0;  // rows of rm_dev/norms_dev currently uploaded
	// Per-layer dirty tracking for the per-chunk adjacency syncs of the
	// GPU build (v3 keeps a padded device adjacency for EVERY layer):
	// flag dedup + append-only id list, so a sync never scans all nelem
	// flags (that scan was O(nelem x chunks) — quadratic at 10M rows).
	// Only active (allocated) inside the GPU snapshot build, where layers
	// are pre-created so h->layers never reallocates under workers.
	struct DirtyTrack {
		std::vector<uint8_t> flag;
		std::vector<uint32_t> list;
		std::atomic<uint32_t> n{0};
	};
	std::vector<std::unique_ptr<DirtyTrack>> dtrack; // indexed by layer
	// flag+append; callers serialize per node (node locks / sequential
	// loops), so the flag test-and-set cannot race for one node
	inline void mark_dirty(uint32_t layer, uint32_t e) {
		if (layer >= dtrack.size() || !dtrack[layer])
			return;
		DirtyTrack &D = *dtrack[layer];
		if (D.flag[e])
			return;
		D.flag[e] = 1;
		D.list[D.n.fetch_add(1, std::memory_order_relaxed)] = e;
	}
	std::string err;
};

namespace hnsw {

static inline const float *vec(const sdbv_hnsw *h, uint32_t id) {
	return h->vecs.data() + (uint64_t)id * h->d;
}

static double dist(const sdbv_hnsw *h, const float *a, double a_norm,
                   uint32_t id) {
	if (h->metric == SDBV_METRIC_COSINE) {
		double dot = host_dot_f32(a, vec(h, id), h->d);
		return 1.0 - dot / (a_norm * h->norms[id]);
	}
	float acc = 0;
	const float *b = vec(h, id);
	for (uint32_t i = 0; i < h->d; i++) {
		float diff = a[i] - b[i];
		acc += diff * diff;
	}
	return sqrt((double)acc);
}
static double dist_ee(const sdbv_hnsw *h, uint32_t a, uint32_t b) {
	return dist(h, vec(h, a), h->metric == SDBV_METRIC_COSINE ? h->norms[a] : 0,
	            b);
}

static std::vector<uint32_t> get_edges(sdbv_hnsw *h, const Layer &layer,
                                       uint32_t id, bool locked) {
	if (!locked)
		return id < layer.edges.size() ? layer.edges[id]
		                               : std::vector<uint32_t>{};
	std::lock_guard<std::mutex> lk(h->node_locks[id & 4095]);
	return id < layer.edges.size() ? layer.edges[id] : std::vector<uint32_t>{};
}

} // namespace hnsw
struct sdbv_index; // defined below (index layer)
namespace hnsw {

// Pending-docs context for index searches (hnsw/index.rs knn path): the
// DocId bitmap from search_pendings plus the element->docs accessor.
struct IdxPend {
	const std::set<uint64_t> *pending;
	const ::sdbv_index *ix;
};
static bool idx_all_docs_pending(const IdxPend *p, uint32_t e_id);

// Epoch-stamped visited set (build hot path): O(1) insert, no per-search
// allocation or clearing. Drop-in for unordered_set<uint32_t> in
// search_layer_host (template).
struct VisitSet {
	std::vector<uint32_t> stamp;
	uint32_t epoch = 0;
	void begin(size_t n) {
		if (stamp.size() < n)
			stamp.resize(n, 0);
		if (++epoch == 0) {
			std::fill(stamp.begin(), stamp.end(), 0);
			epoch = 1;
		}
	}
	// mimics unordered_set::insert().second
	struct R {
		bool second;
	};
	R insert(uint32_t id) {
		if (id >= stamp.size())
			stamp.resize(id + 1, 0);
		if (stamp[id] == epoch)
			return {false};
		stamp[id] = epoch;
		return {true};
	}
};

// layer.rs:184-223 — host-distance variant (build path + index host path).
// `pend`: an element whose docs are ALL pending is excluded from
// `candidates` only; it still enters `w` (layer.rs:209-212, the reference
// pushes to w outside the exclusion check — restated as-is).
template <class VS>
static void search_layer_host(sdbv_hnsw *h, const Layer &layer, const float *q,
                              double q_norm, PQ &candidates, VS &visited,
                              PQ &w, uint32_t ef, bool locked,
                              const IdxPend *pend = nullptr) {
	double fq = w.peek_last_dist(DBL_MAX);
	double cd;
	uint32_t doc;
	std::vector<uint32_t> scratch; // locked-mode edge snapshot (reused)
	while (candidates.pop_first(&cd, &doc)) {
		if (cd > fq)
			break;
		const std::vector<uint32_t> *edges_p;
		if (!locked) {
			static const std::vector<uint32_t> kEmpty;
			edges_p = doc < layer.edges.size() ? &layer.edges[doc] : &kEmpty;
		} else {
			scratch.clear();
			std::lock_guard<std::mutex> lk(h->node_locks[doc & 4095]);
			if (doc < layer.edges.size())
				scratch = layer.edges[doc];
			edges_p = &scratch;
		}
		// prefetch the frontier's vectors: the expansion reads ~m0 random
		// 3 KB rows far larger than any cache level at bench scale
		for (uint32_t e : *edges_p)
			__builtin_prefetch(vec(h, e), 0, 1);
		for (uint32_t e : *edges_p) {
			if (!visited.insert(e).second)
				continue;
			// elements.get_vector -> None for removed elements
			// (layer.rs:206): dangling edges left by insert-time pruning
			// asymmetry are skipped after the visited mark
			if (e < h->elem_present.size() && !h->elem_present[e])
				continue;
			double ed = dist(h, q, q_norm, e);
			if (ed < fq || w.n < ef) {
				if (!pend || !idx_all_docs_pending(pend, e))
					candidates.push(ed, e);
				w.push(ed, e);
				if (w.n > ef)
					w.pop_last();
				fq = w.peek_last_dist(DBL_MAX);
			}
		}
	}
}

// heuristic.rs:193-216
static bool is_closer(sdbv_hnsw *h, double ed, uint32_t e,
                      std::vector<uint32_t> &r) {
	for (uint32_t rid : r)
		if (ed > dist_ee(h, e, rid))
			return false;
	r.push_back(e);
	return true;
}

// heuristic.rs:35-116 (+ extend :118-157; `ignore` = heuristic.rs:130-134,
// the element being removed is excluded from the extension set)
static void select_neighbors(sdbv_hnsw *h, const Layer &layer, uint32_t q_id,
                             const float *q_pt, double q_norm, PQ c,
                             std::vector<uint32_t> &res, bool locked,
                             int64_t ignore = -1) {
	if (h->extend) {
		std::unordered_set<uint32_t> ex;
		auto base = c.to_vec();
		for (auto &e : base)
			ex.insert(e.second);
		if (ignore >= 0)
			ex.insert((uint32_t)ignore);
		for (auto &e : base)
			for (uint32_t adj : get_edges(h, layer, e.second, locked))
				if (adj != q_id && ex.insert(adj).second) {
					// get_distance -> None for removed elements
					if (adj < h->elem_present.size() &&
					    !h->elem_present[adj])
						continue;
					c.push(dist(h, q_pt, q_norm, adj), adj);
				}
	}
	uint32_t m_max = layer.m_max;
	if (c.n <= m_max) {
		for (auto &e : c.to_vec())
			res.push_back(e.second);
		return;
	}
	std::vector<uint32_t> pruned;
	double ed;
	uint32_t e;
	while (c.pop_first(&ed, &e)) {
		if (is_closer(h, ed, e, res)) {
			if (res.size() == m_max)
				break;
		} else if (h->keep) {
			pruned.push_back(e);
		}
	}
	if (h->keep) {
		size_t nmore = m_max - res.size();
		for (size_t i = 0; i < nmore && i < pruned.size(); i++)
			res.push_back(pruned[i]);
	}
}

// layer.rs:342-387
// The insert's apply half (select + bidirectional edges + prunes) over a
// precomputed candidate window `w` — shared by the classic insert (whose
// search ran just now) and the chunked snapshot build (whose search ran
// against the pre-chunk graph; see sdbv_hnsw_insert_batch_snapshot).
static void layer_insert_apply(sdbv_hnsw *h, Layer &layer, uint32_t q_id,
                               const float *q_pt, double q_norm, PQ w,
                               bool locked) {
	// device-adjacency dirty marking for the GPU build's classic-path
	// inserts (no-op unless a build allocated dtrack for this layer);
	// layers never reallocates during builds, so the index is stable
	const uint32_t li = (uint32_t)(&layer - h->layers.data());
	std::vector<uint32_t> neighbors;
	select_neighbors(h, layer, q_id, q_pt, q_norm, std::move(w), neighbors,
	                 locked);
	{
		// append (not overwrite): a concurrent inserter may already have
		// back-linked into q_id; sequential mode this is plain assignment
		std::lock_guard<std::mutex> lk(h->node_locks[q_id & 4095]);
		auto &eq = layer.edges[q_id];
		for (uint32_t e : neighbors)
			if (e != q_id &&
			    std::find(eq.begin(), eq.end(), e) == eq.end())
				eq.push_back(e);
		h->mark_dirty(li, q_id);
	}
	for (uint32_t e : neighbors) {
		if (e == q_id)
			continue;
		std::vector<uint32_t> conn;
		{
			std::lock_guard<std::mutex> lk(h->node_locks[e & 4095]);
			// graph.rs:52-64: the back-edge entry().or_insert IMPLICITLY
			// creates a missing target node (e.g. an upper-layer seed)
			if (e < layer.in_layer.size())
				layer.in_layer[e] = 1;
			auto &ee = layer.edges[e];
			if (std::find(ee.begin(), ee.end(), q_id) == ee.end()) {
				ee.push_back(q_id);
				h->mark_dirty(li, e);
			}
			if (ee.size() > layer.m_max)
				conn = ee;
		}
		if (!conn.empty()) {
			// prune (layer.rs:363-377) — distances computed outside the
			// lock. build_priority_list (layer.rs:389-404) SKIPS removed
			// elements (get_vector -> None), so a prune also cleanses any
			// dangling edges out of e's list.
			PQ ec;
			for (uint32_t nid : conn) {
				if (nid < h->elem_present.size() && !h->elem_present[nid])
					continue;
				ec.push(dist_ee(h, e, nid), nid);
			}
			std::vector<uint32_t> enew;
			select_neighbors(h, layer, e, vec(h, e),
			                 h->metric == SDBV_METRIC_COSINE ? h->norms[e] : 0,
			                 std::move(ec), enew, locked);
			std::lock_guard<std::mutex> lk(h->node_locks[e & 4095]);
			// keep any back-edges another inserter added since the snapshot
			// (parallel mode only; sequential sees none)
			for (uint32_t cur : layer.edges[e])
				if (std::find(conn.begin(), conn.end(), cur) == conn.end() &&
				    std::find(enew.begin(), enew.end(), cur) == enew.end())
					enew.push_back(cur);
			layer.edges[e] = enew;
			h->mark_dirty(li, e);
		}
	}
}

static PQ layer_insert(sdbv_hnsw *h, Layer &layer, uint32_t q_id,
                       const float *q_pt, double q_norm, PQ eps, bool locked) {
	PQ w = eps;
	static thread_local VisitSet visited;
	visited.begin(h->vecs.size() / h->d);
	for (auto &e : eps.to_vec())
		visited.insert(e.second);
	search_layer_host(h, layer, q_pt, q_norm, eps, visited, w, h->efc, locked);
	PQ out = w;
	layer_insert_apply(h, layer, q_id, q_pt, q_norm, std::move(w), locked);
	return out;
}

// hnsw/mod.rs:263-266 (level RNG restatement — same contract as the oracle)
static inline uint64_t splitmix_host(uint64_t z) {
	z += 0x9E3779B97F4A7C15ULL;
	z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
	z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
	return z ^ (z >> 31);
}
static uint32_t next_level(sdbv_hnsw *h) {
	h->rng_state = splitmix_host(h->rng_state);
	double u = (double)(h->rng_state >> 11) * 0x1.0p-53;
	if (u <= 0.0)
		u = 0x1.0p-53;
	double lvl = std::floor(-std::log(u) * h->ml);
	return (uint32_t)std::min(std::max(lvl, 0.0), 30.0);
}

// hnsw/mod.rs:230-394 insert at a given level (vector already appended)
static void insert_at(sdbv_hnsw *h, uint32_t q_id, uint32_t q_level,
                      bool locked) {
	const float *q_pt = vec(h, q_id);
	double q_norm = h->metric == SDBV_METRIC_COSINE ? h->norms[q_id] : 0;
	uint32_t top_up;
	{
		std::lock_guard<std::mutex> lk(h->global_mu);
		top_up = (uint32_t)h->layers.size() - 1;
		for (uint32_t i = top_up; i < q_level; i++)
			h->layers.push_back(hnsw::Layer{
			    std::vector<std::vector<uint32_t>>(h->vecs.size() / h->d),
			    h->m});
		uint64_t nelem = h->vecs.size() / h->d;
		for (auto &l : h->layers) {
			if (l.edges.size() <= q_id)
				l.edges.resize(nelem);
			if (l.in_layer.size() < nelem)
				l.in_layer.resize(nelem, 0);
		}
		// node membership on layers 0..=q_level (graph.rs add_node /
		// add_empty_node in insert_first_element and insert_element)
		for (uint32_t l = 0;
		     l < (uint32_t)h->layers.size() && l <= q_level; l++)
			h->layers[l].in_layer[q_id] = 1;
		h->dirty = true;
		if (h->enter_point < 0) {
			h->enter_point = q_id;
			return;
		}
	}
	uint64_t ep_id = (uint64_t)h->enter_point;
	double ep_dist = dist(h, q_pt, q_norm, (uint32_t)ep_id);
	if (q_level < top_up) {
		for (uint32_t l = top_up; l > q_level; l--) {
			PQ cand;
			cand.push(ep_dist, (uint32_t)ep_id);
			std::unordered_set<uint32_t> visited{(uint32_t)ep_id};
			PQ w = cand;
			search_layer_host(h, h->layers[l], q_pt, q_norm, cand, visited, w,
			                  1, locked);
			double dd;
			uint32_t ii;
			if (w.peek_first(&dd, &ii)) {
				ep_dist = dd;
				ep_id = ii;
			}
		}
	}
	PQ eps;
	eps.push(ep_dist, (uint32_t)ep_id);
	uint32_t ins_to = std::min(q_level, top_up);
	for (uint32_t l = ins_to; l >= 1; l--)
		eps = layer_insert(h, h->layers[l], q_id, q_pt, q_norm, std::move(eps),
		                   locked);
	layer_insert(h, h->layers[0], q_id, q_pt, q_norm, std::move(eps), locked);
	if (q_level > top_up) {
		std::lock_guard<std::mutex> lk(h->global_mu);
		h->enter_point = q_id;
	}
}

// Chunked SNAPSHOT searches (the §8f-rank-3 structure): per chunk, every
// element's efc-search runs against the graph AS OF the chunk start
// (read-only — no locks, embarrassingly parallel; the GPU build batches
// the same searches onto the persistent kernel).

// Upper-layer greedy descent only (host half shared by the CPU and GPU
// snapshot paths); returns the layer-0 entry point + distance.
static void snapshot_descend_one(sdbv_hnsw *h, uint32_t q_id, double q_norm,
                                 uint32_t *ep_out, double *epd_out) {
	const float *q_pt = vec(h, q_id);
	uint32_t ep_id = (uint32_t)h->enter_point;
	double ep_dist = dist(h, q_pt, q_norm, ep_id);
	for (size_t l = h->layers.size() - 1; l >= 1; l--) {
		PQ cand;
		cand.push(ep_dist, ep_id);
		std::unordered_set<uint32_t> visited{ep_id};
		PQ w = cand;
		search_layer_host(h, h->layers[l], q_pt, q_norm, cand, visited, w,
		                  1, false);
		double dd;
		uint32_t ii;
		if (w.peek_first(&dd, &ii)) {
			ep_dist = dd;
			ep_id = ii;
		}
	}
	*ep_out = ep_id;
	*epd_out = ep_dist;
}

static void snapshot_search_one(sdbv_hnsw *h, uint32_t q_id, PQ &w_out) {
	const float *q_pt = vec(h, q_id);
	double q_norm = h->metric == SDBV_METRIC_COSINE ? h->norms[q_id] : 0;
	uint32_t ep_id;
	double ep_dist;
	snapshot_descend_one(h, q_id, q_norm, &ep_id, &ep_dist);
	PQ eps;
	eps.push(ep_dist, ep_id);
	PQ w = eps;
	static thread_local VisitSet visited;
	visited.begin(h->vecs.size() / h->d);
	visited.insert(ep_id);
	search_layer_host(h, h->layers[0], q_pt, q_norm, eps, visited, w,
	                  h->efc, false);
	w_out = std::move(w);
}

// ef-search at ANY layer from a given eps window (the insert path's
// multi-ep seeding, layer.rs:342-358) against the snapshot graph.
static void snapshot_search_layer_eps(sdbv_hnsw *h, uint32_t layer_idx,
                                      uint32_t q_id, const PQ &eps_in,
                                      uint32_t ef, PQ &w_out) {
	const float *q_pt = vec(h, q_id);
	double q_norm = h->metric == SDBV_METRIC_COSINE ? h->norms[q_id] : 0;
	PQ eps = eps_in;
	PQ w = eps;
	static thread_local VisitSet visited;
	visited.begin(h->vecs.size() / h->d);
	for (auto &e : eps_in.to_vec())
		visited.insert(e.second);
	search_layer_host(h, h->layers[layer_idx], q_pt, q_norm, eps, visited,
	                  w, ef, false);
	w_out = std::move(w);
}

// Greedy descent through layers top..2 only (ef=1; the layer-1 hop is the
// batched builds' device work).
static void descend_to_layer2(sdbv_hnsw *h, uint32_t q_id, double q_norm,
                              uint32_t *ep_out, double *epd_out) {
	const float *q_pt = vec(h, q_id);
	uint32_t ep_id = (uint32_t)h->enter_point;
	double ep_dist = dist(h, q_pt, q_norm, ep_id);
	for (size_t l = h->layers.size() - 1; l >= 2; l--) {
		PQ cand;
		cand.push(ep_dist, ep_id);
		std::unordered_set<uint32_t> visited{ep_id};
		PQ w = cand;
		search_layer_host(h, h->layers[l], q_pt, q_norm, cand, visited, w,
		                  1, false);
		double dd;
		uint32_t ii;
		if (w.peek_first(&dd, &ii)) {
			ep_dist = dd;
			ep_id = ii;
		}
	}
	*ep_out = ep_id;
	*epd_out = ep_dist;
}

// ---- batched per-layer apply (snapshot-build v3) ----
// The same select/prune algorithm as layer_insert_apply, but in three
// bulk phases per chunk AND PER LAYER: (A) every element's neighbour
// select against the post-search graph, read-only and parallel; (B) all
// edge appends, in element order, sequential (deterministic and cheap);
// (C) one prune pass over every node that ended over m_max, parallel
// (prunes are independent: enew ⊆ conn, no cascading). This is ONE valid
// serialization of the parallel interleaved apply — same algorithm, a
// fixed schedule — and the form whose distance work (the RAM-bound part)
// batches onto the device in the GPU build. Quality is pinned by the same
// recall bars; the GPU twin must match this host twin bit-exactly.
struct ApplyItem {
	uint32_t q_id;
	PQ w;
	std::vector<uint32_t> neighbors; // phase-A output
};

static std::vector<uint32_t> batched_apply_phaseB(sdbv_hnsw *h,
                                                  uint32_t layer_idx,
                                                  std::vector<ApplyItem *> &it);
static void batched_apply_phaseC_host(sdbv_hnsw *h, uint32_t layer_idx,
                                      const std::vector<uint32_t> &overfull,
                                      int nthreads);

// host phase A + B + C (the twin's apply; the GPU build replaces A and C
// with k_pair_mats/k_heur_select)
static void batched_apply_host(sdbv_hnsw *h, uint32_t layer_idx,
                               std::vector<ApplyItem *> &items,
                               int nthreads) {
	Layer &L = h->layers[layer_idx];
	// phase A: selects (read-only graph)
	{
		std::atomic<uint64_t> cursor{0};
		auto worker = [&]() {
			uint64_t j;
			while ((j = cursor.fetch_add(1)) < items.size()) {
				ApplyItem &it = *items[j];
				const float *q_pt = vec(h, it.q_id);
				double q_norm = h->metric == SDBV_METRIC_COSINE
				                    ? h->norms[it.q_id]
				                    : 0;
				it.neighbors.clear();
				select_neighbors(h, L, it.q_id, q_pt, q_norm, it.w,
				                 it.neighbors, false);
			}
		};
		std::vector<std::thread> ts;
		int nt = std::max(1, std::min<int>(nthreads, (int)items.size()));
		for (int t = 1; t < nt; t++)
			ts.emplace_back(worker);
		worker();
		for (auto &t : ts)
			t.join();
	}
	auto overfull = batched_apply_phaseB(h, layer_idx, items);
	batched_apply_phaseC_host(h, layer_idx, overfull, nthreads);
}

// phase B: appends, element order (graph.rs:52-64 entry().or_insert —
// including the implicit creation of a missing back-edge target, e.g. an
// eps node from the layer above that is not a member of THIS layer);
// returns the deduped list of nodes that ended over m_max (prune targets)
static std::vector<uint32_t> batched_apply_phaseB(sdbv_hnsw *h,
                                                  uint32_t layer_idx,
                                                  std::vector<ApplyItem *> &items) {
	Layer &L = h->layers[layer_idx];
	std::vector<uint32_t> overfull;
	for (auto *itp : items) {
		auto &it = *itp;
		auto &eq = L.edges[it.q_id];
		for (uint32_t e : it.neighbors)
			if (e != it.q_id &&
			    std::find(eq.begin(), eq.end(), e) == eq.end())
				eq.push_back(e);
		L.in_layer[it.q_id] = 1;
		h->mark_dirty(layer_idx, it.q_id);
	}
	for (auto *itp : items) {
		auto &it = *itp;
		for (uint32_t e : it.neighbors) {
			if (e == it.q_id)
				continue;
			auto &ee = L.edges[e];
			if (e < L.in_layer.size())
				L.in_layer[e] = 1;
			if (std::find(ee.begin(), ee.end(), it.q_id) == ee.end()) {
				ee.push_back(it.q_id);
				if (ee.size() > L.m_max)
					overfull.push_back(e); // deduped below
				h->mark_dirty(layer_idx, e);
			}
		}
	}
	std::sort(overfull.begin(), overfull.end());
	overfull.erase(std::unique(overfull.begin(), overfull.end()),
	               overfull.end());
	return overfull;
}

// phase C: prunes (layer.rs:363-377), parallel over distinct nodes
static void batched_apply_phaseC_host(sdbv_hnsw *h, uint32_t layer_idx,
                                      const std::vector<uint32_t> &overfull,
                                      int nthreads) {
	Layer &L = h->layers[layer_idx];
	std::atomic<uint64_t> cursor{0};
	auto worker = [&]() {
		uint64_t i;
		while ((i = cursor.fetch_add(1)) < overfull.size()) {
			uint32_t e = overfull[i];
			const auto conn = L.edges[e]; // copy (read-only source)
			PQ ec;
			for (uint32_t nid : conn) {
				if (nid < h->elem_present.size() &&
				    !h->elem_present[nid])
					continue;
				ec.push(dist_ee(h, e, nid), nid);
			}
			std::vector<uint32_t> enew;
			select_neighbors(h, L, e, vec(h, e),
			                 h->metric == SDBV_METRIC_COSINE
			                     ? h->norms[e]
			                     : 0,
			                 std::move(ec), enew, false);
			L.edges[e] = enew;
			h->mark_dirty(layer_idx, e);
		}
	};
	std::vector<std::thread> ts;
	int nt = std::max(1, std::min<int>(nthreads,
	                                   (int)std::max<size_t>(overfull.size(),
	                                                         1)));
	for (int t = 1; t < nt; t++)
		ts.emplace_back(worker);
	worker();
	for (auto &t : ts)
		t.join();
}

// ---- graph element removal (sequential only: apply_pendings holds the
// reference's write lock; the parallel bench-mode build never removes) ----

// layer.rs:92-108 search_single_with_ignore: seeded FROM the ignored
// element; returns the closest found element or -1 (None).
static int64_t search_single_with_ignore(sdbv_hnsw *h, const Layer &layer,
                                         const float *pt, double pt_norm,
                                         uint32_t ignore_id, uint32_t ef) {
	std::unordered_set<uint32_t> visited{ignore_id};
	PQ candidates;
	candidates.push(dist(h, pt, pt_norm, ignore_id), ignore_id);
	PQ w;
	search_layer_host(h, layer, pt, pt_norm, candidates, visited, w, ef,
	                  false);
	double dd;
	uint32_t ii;
	if (w.peek_first(&dd, &ii))
		return (int64_t)ii;
	return -1;
}

// layer.rs:164-181 search_multi_with_ignore.
static PQ search_multi_with_ignore(sdbv_hnsw *h, const Layer &layer,
                                   const float *pt, double pt_norm,
                                   const std::vector<uint32_t> &ignore_ids,
                                   uint32_t efc) {
	PQ candidates;
	for (uint32_t id : ignore_ids)
		candidates.push(dist(h, pt, pt_norm, id), id);
	std::unordered_set<uint32_t> visited(ignore_ids.begin(),
	                                     ignore_ids.end());
	PQ w;
	search_layer_host(h, layer, pt, pt_norm, candidates, visited, w, efc,
	                  false);
	return w;
}

// layer.rs:408-460 HnswLayer::remove: drop node + back-edges, then repair
// each former neighbour (efc-search ignoring {q_id, e_id}, heuristic
// re-selection with ignore=e_id, one-directional set_node).
static bool layer_remove(sdbv_hnsw *h, Layer &layer, uint32_t e_id) {
	if (!layer.has(e_id))
		return false;
	std::vector<uint32_t> f_ids = std::move(layer.edges[e_id]);
	layer.edges[e_id].clear();
	layer.in_layer[e_id] = 0;
	for (uint32_t f : f_ids) {
		auto &fe = layer.edges[f];
		fe.erase(std::remove(fe.begin(), fe.end(), e_id), fe.end());
	}
	for (uint32_t q_id : f_ids) {
		const float *q_pt = vec(h, q_id);
		double q_norm =
		    h->metric == SDBV_METRIC_COSINE ? h->norms[q_id] : 0;
		PQ c = search_multi_with_ignore(h, layer, q_pt, q_norm,
		                                {q_id, e_id}, h->efc);
		std::vector<uint32_t> q_new_conn;
		select_neighbors(h, layer, q_id, q_pt, q_norm, std::move(c),
		                 q_new_conn, false, (int64_t)e_id);
		layer.edges[q_id] = q_new_conn; // graph.set_node
	}
	return true;
}

// hnsw/mod.rs:398-455 Hnsw::remove.
static bool hnsw_remove(sdbv_hnsw *h, uint32_t e_id) {
	if (e_id >= h->next_id || !h->elem_present[e_id])
		return false; // elements.get_vector -> None
	bool removed = false;
	const float *e_pt = vec(h, e_id);
	double e_norm = h->metric == SDBV_METRIC_COSINE ? h->norms[e_id] : 0;
	int64_t new_ep = ((int64_t)e_id == h->enter_point) ? -1 : h->enter_point;
	for (size_t l = h->layers.size() - 1; l >= 1; l--) {
		if (new_ep < 0)
			new_ep = search_single_with_ignore(h, h->layers[l], e_pt,
			                                   e_norm, e_id, h->efc);
		if (layer_remove(h, h->layers[l], e_id))
			removed = true;
	}
	if (new_ep < 0)
		new_ep = search_single_with_ignore(h, h->layers[0], e_pt, e_norm,
		                                   e_id, h->efc);
	if (layer_remove(h, h->layers[0], e_id))
		removed = true;
	h->elem_present[e_id] = 0; // elements.remove
	h->enter_point = new_ep;
	h->dirty = true;
	return removed;
}

} // namespace hnsw

extern "C" {

// Host-side synthetic generator (same committed bit contract as k_gen_cm and
// the oracle); parallel over rows. Input prep for benches/tests — no GPU.
void sdbv_gen_f32(uint64_t seed, uint64_t row0, uint64_t nrows, uint32_t d,
                  float *out) {
	int nthreads = (int)std::thread::hardware_concurrency();
	if (nthreads < 1)
		nthreads = 1;
	if (nrows < 4096)
		nthreads = 1;
	std::vector<std::thread> ws;
	std::atomic<uint64_t> next{0};
	auto work = [&] {
		for (;;) {
			uint64_t i = next.fetch_add(4096);
			if (i >= nrows)
				break;
			uint64_t end = std::min(nrows, i + 4096);
			for (uint64_t r = i; r < end; r++)
				for (uint32_t j = 0; j < d; j++)
					out[r * d + j] =
					    d_gen_elem(seed, (row0 + r) * (uint64_t)d + j);
		}
	};
	for (int w = 0; w < nthreads - 1; w++)
		ws.emplace_back(work);
	work();
	for (auto &w : ws)
		w.join();
}

int sdbv_hnsw_create(sdbv_ctx *ctx, uint32_t d, uint8_t metric, uint32_t m,
                     uint32_t m0, uint32_t efc, int extend, int keep,
                     uint64_t seed, double ml, sdbv_hnsw **out) {
	// ctx may be NULL for a host-only (build/export) index; finalize and
	// knn require a context and fail loudly without one.
	if (d == 0 || d > MAX_D || (d % 4) != 0 ||
	    metric > SDBV_METRIC_EUCLIDEAN || m == 0 || m0 == 0)
		return SDBV_ERR_BAD_ARG;
	auto *h = new sdbv_hnsw();
	h->ctx = ctx;
	h->d = d;
	h->metric = metric;
	h->m = m;
	h->m0 = m0;
	h->efc = efc;
	h->extend = extend != 0;
	h->keep = keep != 0;
	h->ml = ml;
	h->rng_state = seed;
	h->layers.push_back(hnsw::Layer{{}, m0});
	*out = h;
	return SDBV_OK;
}

void sdbv_hnsw_destroy(sdbv_hnsw *h) {
	if (!h)
		return;
	if (h->rows_dev)
		(void)hipFree(h->rows_dev);
	if (h->dout_dev)
		(void)hipFree(h->dout_dev);
	if (h->q_dev)
		(void)hipFree(h->q_dev);
	if (h->rows_pinned)
		(void)hipHostFree(h->rows_pinned);
	if (h->dists_pinned)
		(void)hipHostFree(h->dists_pinned);
	for (void *p : {(void *)h->rm_dev, (void *)h->offsets_dev,
	                (void *)h->edges_dev, (void *)h->norms_dev,
	                (void *)h->vis_dev})
		if (p)
			(void)hipFree(p);
	delete h;
}

static void hnsw_free_device_state(sdbv_hnsw *h); // defined below finalize

static void hnsw_append_vec(sdbv_hnsw *h, const float *pt) {
	h->vecs.insert(h->vecs.end(), pt, pt + h->d);
	h->elem_present.push_back(1);
	if (h->metric == SDBV_METRIC_COSINE)
		h->norms.push_back(sqrt(hnsw::host_sumsq_f32(pt, h->d)));
}

int sdbv_hnsw_insert(sdbv_hnsw *h, const float *pt) {
	if (!h)
		return SDBV_ERR_BAD_ARG;
	// Writes on a finalized graph invalidate the device copy (same contract
	// as hnsw_remove): drop it and let the next search re-finalize. The
	// round-1 BAD_ARG guard here made idx_vd_insert silently skip the graph
	// insert for every update applied after a GPU search (driver GPUTEST_r01
	// failure: updated doc missing from post-apply top-K).
	if (h->finalized) {
		hnsw_free_device_state(h);
		h->dirty = true;
	}
	uint32_t q_id = (uint32_t)h->next_id++;
	hnsw_append_vec(h, pt);
	hnsw::insert_at(h, q_id, hnsw::next_level(h), /*locked=*/false);
	return SDBV_OK;
}

int sdbv_hnsw_insert_batch(sdbv_hnsw *h, const float *pts, uint64_t n,
                           int nthreads) {
	if (!h || h->finalized)
		return SDBV_ERR_BAD_ARG;
	if (nthreads <= 0)
		nthreads = (int)std::thread::hardware_concurrency();
	if (nthreads <= 1) {
		for (uint64_t i = 0; i < n; i++) {
			int rc = sdbv_hnsw_insert(h, pts + i * h->d);
			if (rc)
				return rc;
		}
		return SDBV_OK;
	}
	uint64_t base = h->next_id;
	// levels drawn deterministically per ordinal (sequential RNG)
	std::vector<uint32_t> levels(n);
	for (uint64_t i = 0; i < n; i++)
		levels[i] = hnsw::next_level(h);
	h->vecs.reserve(h->vecs.size() + n * h->d);
	for (uint64_t i = 0; i < n; i++)
		hnsw_append_vec(h, pts + i * h->d);
	h->next_id += n;
	// pre-size every layer to the final element count (no growth races)
	{
		uint32_t maxl = 0;
		for (auto l : levels)
			maxl = std::max(maxl, l);
		std::lock_guard<std::mutex> lk(h->global_mu);
		while (h->layers.size() <= maxl)
			h->layers.push_back(hnsw::Layer{{}, h->m});
		for (auto &l : h->layers)
			l.edges.resize(h->next_id);
	}
	// the first elements (empty/near-empty graph) go in alone so every
	// worker sees a connected entry point
	uint64_t start = 0;
	uint64_t warm = std::min<uint64_t>(n, h->enter_point < 0 ? 64 : 0);
	for (; start < warm; start++)
		hnsw::insert_at(h, (uint32_t)(base + start), levels[start], true);
	std::atomic<uint64_t> next{start};
	std::vector<std::thread> workers;
	for (int w = 0; w < nthreads; w++)
		workers.emplace_back([&] {
			for (;;) {
				uint64_t i = next.fetch_add(1);
				if (i >= n)
					break;
				hnsw::insert_at(h, (uint32_t)(base + i), levels[i], true);
			}
		});
	for (auto &w : workers)
		w.join();
	// Promote the enter point to a max-level element of this batch: with
	// pre-created layers insert_at's q_level > top_up promotion never
	// fires, which would leave the upper layers unreachable from a
	// level-0 enter point (greedy descents would pass through them).
	{
		uint32_t maxl = 0;
		for (auto l : levels)
			maxl = std::max(maxl, l);
		std::lock_guard<std::mutex> lk(h->global_mu);
		uint32_t cur_top = (uint32_t)h->layers.size() - 1;
		if (h->enter_point >= 0 && maxl >= cur_top &&
		    !h->layers[cur_top].has((uint32_t)h->enter_point)) {
			for (uint64_t i = 0; i < n; i++)
				if (levels[i] == maxl) {
					h->enter_point = (int64_t)(base + i);
					break;
				}
		}
	}
	return SDBV_OK;
}

// Chunked snapshot build (see hnsw::snapshot_search_one above):
int sdbv_hnsw_insert_batch_snapshot(sdbv_hnsw *h, const float *pts,
                                    uint64_t n, uint32_t chunk,
                                    int nthreads) {
	using namespace hnsw;
	if (!h || h->finalized || chunk == 0)
		return SDBV_ERR_BAD_ARG;
	if (nthreads <= 0)
		nthreads = (int)std::thread::hardware_concurrency();
	uint64_t base = h->next_id;
	std::vector<uint32_t> levels(n);
	for (uint64_t i = 0; i < n; i++)
		levels[i] = next_level(h); // sequential RNG contract, per ordinal
	h->next_id += n;
	for (uint64_t i = 0; i < n; i++)
		hnsw_append_vec(h, pts + i * h->d);
	// pre-create every layer the batch will need and size them up front:
	// concurrent insert_at must never grow h->layers (vector reallocation
	// under readers — same discipline as the classic parallel batch path)
	{
		std::lock_guard<std::mutex> lk(h->global_mu);
		uint32_t max_level = 0;
		for (uint64_t i = 0; i < n; i++)
			max_level = std::max(max_level, levels[i]);
		while (h->layers.size() <= max_level)
			h->layers.push_back(hnsw::Layer{{}, h->m});
		uint64_t nelem = h->vecs.size() / h->d;
		for (auto &l : h->layers) {
			if (l.edges.size() < nelem)
				l.edges.resize(nelem);
			if (l.in_layer.size() < nelem)
				l.in_layer.resize(nelem, 0);
		}
	}
	for (uint64_t c0 = 0; c0 < n; c0 += chunk) {
		uint64_t c1 = std::min(n, c0 + chunk);
		// upper-level elements first (~1/m of the chunk) so the layer
		// structure exists before the snapshot searches; inserted with the
		// classic parallel path (striped locks) — the first element of an
		// empty graph goes alone to establish the enter point
		std::vector<uint64_t> upper, flat;
		for (uint64_t i = c0; i < c1; i++) {
			if (h->enter_point < 0)
				insert_at(h, (uint32_t)(base + i), levels[i], false);
			else if (levels[i] > 0)
				upper.push_back(i);
			else
				flat.push_back(i);
		}
		if (!upper.empty()) {
			std::atomic<uint64_t> ucursor{0};
			auto upper_worker = [&]() {
				uint64_t j;
				while ((j = ucursor.fetch_add(1)) < upper.size())
					insert_at(h, (uint32_t)(base + upper[j]),
					          levels[upper[j]], true);
			};
			std::vector<std::thread> ts;
			int nt = std::max(1, std::min<int>(nthreads,
			                                   (int)upper.size()));
			for (int t = 1; t < nt; t++)
				ts.emplace_back(upper_worker);
			upper_worker();
			for (auto &t : ts)
				t.join();
		}
		// snapshot searches: read-only graph, parallel, no locks
		std::vector<PQ> ws(flat.size());
		std::atomic<uint64_t> cursor{0};
		auto search_worker = [&]() {
			uint64_t j;
			while ((j = cursor.fetch_add(1)) < flat.size())
				snapshot_search_one(h, (uint32_t)(base + flat[j]), ws[j]);
		};
		{
			std::vector<std::thread> ts;
			int nt = std::max(1, std::min<int>(nthreads,
			                                   (int)flat.size()));
			for (int t = 1; t < nt; t++)
				ts.emplace_back(search_worker);
			search_worker();
			for (auto &t : ts)
				t.join();
		}
		// apply phase: striped node locks, parallel (the parallel build's
		// existing concurrency contract)
		for (uint64_t j : flat)
			h->layers[0].in_layer[base + j] = 1;
		std::atomic<uint64_t> acursor{0};
		auto apply_worker = [&]() {
			uint64_t j;
			while ((j = acursor.fetch_add(1)) < flat.size()) {
				uint32_t q_id = (uint32_t)(base + flat[j]);
				const float *q_pt = vec(h, q_id);
				double q_norm = h->metric == SDBV_METRIC_COSINE
				                    ? h->norms[q_id]
				                    : 0;
				layer_insert_apply(h, h->layers[0], q_id, q_pt, q_norm,
				                   std::move(ws[j]), true);
			}
		};
		{
			std::vector<std::thread> ts;
			int nt = std::max(1, std::min<int>(nthreads,
			                                   (int)flat.size()));
			for (int t = 1; t < nt; t++)
				ts.emplace_back(apply_worker);
			apply_worker();
			for (auto &t : ts)
				t.join();
		}
	}
	h->dirty = true;
	return SDBV_OK;
}

// Enter-point reachability fix shared by the batch builds (the classic
// parallel build's promotion, generalized): with pre-created layers the
// insert-time promotion never fires, which can leave the upper layers
// unreachable from a level-0 enter point; promote any top-layer member.
static void hnsw_promote_ep(sdbv_hnsw *h) {
	using namespace hnsw;
	if (h->enter_point < 0 || h->layers.size() < 2)
		return;
	std::lock_guard<std::mutex> lk(h->global_mu);
	uint32_t top = (uint32_t)h->layers.size() - 1;
	// the top non-empty layer
	while (top >= 1) {
		const auto &in = h->layers[top].in_layer;
		if (h->layers[top].has((uint32_t)h->enter_point))
			return;
		for (uint64_t i = 0; i < in.size(); i++)
			if (in[i]) {
				h->enter_point = (int64_t)i;
				return;
			}
		top--; // layer empty (pre-created): look lower
	}
}

// Host twin of the GPU batched-apply snapshot build ("snapshot2", v3):
// the full reference insert algorithm with a fixed batched schedule at
// EVERY layer. Per chunk, top-down through the layers: each element
// either greedy-descends (ef=1, layers above its level — search_ep,
// hnsw/mod.rs:521-548) or runs the efc-search whose w both seeds the
// next layer (layer.rs:358's cascade) and feeds the batched apply
// (select/append/prune) at that layer; layer 0 takes every element.
// Searches see the graph as of the chunk start (snapshot semantics at
// every layer — the same chunk/n quality relaxation the v1 snapshot
// build pinned with recall bars). This is the bit-exact CPU reference
// for sdbv_hnsw_insert_batch_snapshot_gpu.
int sdbv_hnsw_insert_batch_snapshot2(sdbv_hnsw *h, const float *pts,
                                     uint64_t n, uint32_t chunk,
                                     int nthreads) {
	using namespace hnsw;
	if (!h || h->finalized || chunk == 0 || n == 0)
		return SDBV_ERR_BAD_ARG;
	if (nthreads <= 0)
		nthreads = (int)std::thread::hardware_concurrency();
	uint64_t base = h->next_id;
	std::vector<uint32_t> levels(n);
	for (uint64_t i = 0; i < n; i++)
		levels[i] = next_level(h); // sequential RNG contract, per ordinal
	h->next_id += n;
	h->vecs.reserve(h->vecs.size() + n * h->d);
	for (uint64_t i = 0; i < n; i++)
		hnsw_append_vec(h, pts + i * h->d);
	{
		std::lock_guard<std::mutex> lk(h->global_mu);
		uint32_t max_level = 0;
		for (uint64_t i = 0; i < n; i++)
			max_level = std::max(max_level, levels[i]);
		while (h->layers.size() <= max_level)
			h->layers.push_back(hnsw::Layer{{}, h->m});
		uint64_t ne = h->vecs.size() / h->d;
		for (auto &l : h->layers) {
			if (l.edges.size() < ne)
				l.edges.resize(ne);
			if (l.in_layer.size() < ne)
				l.in_layer.resize(ne, 0);
		}
	}
	const uint32_t top = (uint32_t)h->layers.size() - 1;
	(void)top;
	// warm-up (the classic parallel build's bootstrap): while the graph is
	// near-empty every batched element can only select the enter point,
	// whose single prune then orphans most of them — the first elements
	// go in classically so batching starts on a connected graph
	const uint64_t warm_until =
	    h->enter_point < 0 ? base + 64 : 0;
	for (uint64_t c0 = 0; c0 < n; c0 += chunk) {
		uint64_t c1 = std::min(n, c0 + chunk);
		std::vector<ApplyItem> items;
		std::vector<PQ> eps_of;
		std::vector<uint32_t> lvl_of;
		std::vector<std::pair<uint32_t, uint32_t>> upper2; // (q_id, level)
		for (uint64_t i = c0; i < c1; i++) {
			if (h->enter_point < 0 || base + i < warm_until) {
				insert_at(h, (uint32_t)(base + i), levels[i], false);
				continue;
			}
			if (levels[i] >= 2) {
				// levels >= 2 (0.4% of elements): FULL classic inserts —
				// progressive at every layer. Half-inserted elements must
				// never be search-visible: a descent terminating on a
				// node without lower-layer edges gives a degenerate ep
				// (measured: out-degree-1 orphans, recall cliff)
				upper2.push_back({(uint32_t)(base + i), levels[i]});
				continue;
			}
			items.push_back(ApplyItem{(uint32_t)(base + i), PQ{}, {}});
			eps_of.emplace_back();
			lvl_of.push_back(levels[i]);
		}
		if (!upper2.empty()) {
			std::atomic<uint64_t> cursor{0};
			auto worker = [&]() {
				uint64_t j;
				while ((j = cursor.fetch_add(1)) < upper2.size())
					insert_at(h, upper2[j].first, upper2[j].second, true);
			};
			std::vector<std::thread> ts;
			int nt = std::max(1,
			                  std::min<int>(nthreads, (int)upper2.size()));
			for (int t = 1; t < nt; t++)
				ts.emplace_back(worker);
			worker();
			for (auto &t : ts)
				t.join();
		}
		if (items.empty())
			continue;
		// levels <= 1: greedy descents through layers top..2 (read-only)
		{
			std::atomic<uint64_t> cursor{0};
			auto worker = [&]() {
				uint64_t j;
				while ((j = cursor.fetch_add(1)) < items.size()) {
					uint32_t q_id = items[j].q_id;
					double qn = h->metric == SDBV_METRIC_COSINE
					                ? h->norms[q_id]
					                : 0;
					uint32_t ep;
					double epd;
					descend_to_layer2(h, q_id, qn, &ep, &epd);
					eps_of[j].push(epd, ep);
				}
			};
			std::vector<std::thread> ts;
			int nt = std::max(1,
			                  std::min<int>(nthreads, (int)items.size()));
			for (int t = 1; t < nt; t++)
				ts.emplace_back(worker);
			worker();
			for (auto &t : ts)
				t.join();
		}
		// layer 1 (batched): search phase (snapshot, parallel), then the
		// apply for every element with level >= 1
		if (h->layers.size() >= 2) {
			{
				std::atomic<uint64_t> cursor{0};
				auto worker = [&]() {
					uint64_t j;
					while ((j = cursor.fetch_add(1)) < items.size()) {
						uint32_t q_id = items[j].q_id;
						if (lvl_of[j] >= 1) {
							snapshot_search_layer_eps(
							    h, 1, q_id, eps_of[j], h->efc,
							    items[j].w);
							eps_of[j] = items[j].w; // cascade
						} else {
							PQ w;
							snapshot_search_layer_eps(
							    h, 1, q_id, eps_of[j], 1, w);
							double dd;
							uint32_t ii;
							if (w.peek_first(&dd, &ii)) {
								PQ ne2;
								ne2.push(dd, ii);
								eps_of[j] = std::move(ne2);
							}
						}
					}
				};
				std::vector<std::thread> ts;
				int nt = std::max(
				    1, std::min<int>(nthreads, (int)items.size()));
				for (int t = 1; t < nt; t++)
					ts.emplace_back(worker);
				worker();
				for (auto &t : ts)
					t.join();
			}
			std::vector<ApplyItem *> ins;
			for (uint64_t j = 0; j < items.size(); j++)
				if (lvl_of[j] >= 1)
					ins.push_back(&items[j]);
			if (!ins.empty())
				batched_apply_host(h, 1, ins, nthreads);
		}
		// layer 0: every element
		{
			std::atomic<uint64_t> cursor{0};
			auto worker = [&]() {
				uint64_t j;
				while ((j = cursor.fetch_add(1)) < items.size())
					snapshot_search_layer_eps(h, 0, items[j].q_id,
					                          eps_of[j], h->efc,
					                          items[j].w);
			};
			std::vector<std::thread> ts;
			int nt = std::max(1,
			                  std::min<int>(nthreads, (int)items.size()));
			for (int t = 1; t < nt; t++)
				ts.emplace_back(worker);
			worker();
			for (auto &t : ts)
				t.join();
		}
		std::vector<ApplyItem *> all;
		for (auto &it : items)
			all.push_back(&it);
		batched_apply_host(h, 0, all, nthreads);
	}
	hnsw_promote_ep(h);
	h->dirty = true;
	return SDBV_OK;
}

// GPU-accelerated chunked snapshot build, v3 (the
// configs[2] 10M-row build). Bit-identical to the host twin
// sdbv_hnsw_insert_batch_snapshot2 (same batched schedule at EVERY layer,
// same restated distance chains), with all RAM-bound work on the device:
//  - per layer, top-down: one ef=1 multi-query launch carries the greedy
//    descents and one efc launch the insert searches (multi-ep seeding =
//    the layer-above w cascade), against per-layer padded device
//    adjacencies kept in sync by k_adj_scatter deltas;
//  - neighbour selects and prunes: k_pair_mats + k_heur_select (the
//    exact heuristic) at every layer.
// The host keeps the deterministic phase-B edge appends and the queue
// bookkeeping. The round-2 10M trace showed the HOST upper-layer inserts
// of v2 were the remaining wall (~110 s per 1M elements); v3 moves them
// here.
int sdbv_hnsw_insert_batch_snapshot_gpu(sdbv_hnsw *h, const float *pts,
                                        uint64_t n, uint32_t chunk,
                                        int nthreads) {
	using namespace hnsw;
	if (!h || !h->ctx || chunk == 0 || n == 0 || h->efc == 0 ||
	    h->efc > HQ_EF_CAP)
		return SDBV_ERR_BAD_ARG;
	if (h->extend || h->m0 > 64 || h->m > 64 || h->efc + 1 > HSEL_CAP)
		// extend-candidates (or oversized select windows) stay on the
		// host twin — same algorithm, host distances
		return sdbv_hnsw_insert_batch_snapshot2(h, pts, n, chunk, nthreads);
	if (h->finalized) { // writes invalidate the finalized device state
		hnsw_free_device_state(h);
		h->dirty = true;
	}
	if (nthreads <= 0)
		nthreads = (int)std::thread::hardware_concurrency();
	sdbv_ctx *ctx = h->ctx;
	std::lock_guard<std::mutex> ctxlk(ctx->mu);
	uint64_t base = h->next_id;
	std::vector<uint32_t> levels(n);
	for (uint64_t i = 0; i < n; i++)
		levels[i] = next_level(h); // sequential RNG contract, per ordinal
	h->next_id += n;
	h->vecs.reserve(h->vecs.size() + n * h->d);
	for (uint64_t i = 0; i < n; i++)
		hnsw_append_vec(h, pts + i * h->d);
	{
		std::lock_guard<std::mutex> lk(h->global_mu);
		uint32_t max_level = 0;
		for (uint64_t i = 0; i < n; i++)
			max_level = std::max(max_level, levels[i]);
		while (h->layers.size() <= max_level)
			h->layers.push_back(hnsw::Layer{{}, h->m});
		uint64_t ne = h->vecs.size() / h->d;
		for (auto &l : h->layers) {
			if (l.edges.size() < ne)
				l.edges.resize(ne);
			if (l.in_layer.size() < ne)
				l.in_layer.resize(ne, 0);
		}
	}
	const uint64_t nelem = h->vecs.size() / h->d;
	const uint32_t d = h->d;
	const uint32_t efc = h->efc;
	const uint32_t nlayers = (uint32_t)h->layers.size();
	// ---- device state: vectors + per-layer padded adjacencies ----
	for (void **p : {(void **)&h->rm_dev, (void **)&h->norms_dev,
	                 (void **)&h->adj_dev, (void **)&h->deg_dev})
		if (*p) {
			(void)hipFree(*p);
			*p = nullptr;
		}
	HIP_CHECK(ctx, hipMalloc(&h->rm_dev, nelem * d * sizeof(float)));
	HIP_CHECK(ctx, hipMemcpy(h->rm_dev, h->vecs.data(),
	                         nelem * d * sizeof(float),
	                         hipMemcpyHostToDevice));
	if (h->metric == SDBV_METRIC_COSINE) {
		HIP_CHECK(ctx, hipMalloc(&h->norms_dev, nelem * sizeof(double)));
		HIP_CHECK(ctx, hipMemcpy(h->norms_dev, h->norms.data(),
		                         nelem * sizeof(double),
		                         hipMemcpyHostToDevice));
	}
	// layer 0 lives in h->adj_dev/deg_dev; layers >= 1 in build-local
	// arrays (all freed at the end; finalize re-exports the CSR)
	// device adjacency for layers 0 and 1 only: levels >= 2 stay on the
	// classic host path (tiny skeleton layers, progressive for quality)
	const uint32_t ndev_layers = std::min<uint32_t>(nlayers, 2);
	std::vector<uint32_t *> adjL(ndev_layers, nullptr),
	    degL(ndev_layers, nullptr);
	std::vector<uint32_t> strideL(ndev_layers);
	std::vector<uint8_t> has_members(ndev_layers, 0);
	h->dtrack.clear();
	h->dtrack.resize(ndev_layers);
	for (uint32_t l = 0; l < ndev_layers; l++) {
		strideL[l] = (((l == 0 ? h->m0 : h->m) + 8) + 7) & ~7u;
		HIP_CHECK(ctx, hipMalloc(&adjL[l],
		                         nelem * (uint64_t)strideL[l] *
		                             sizeof(uint32_t)));
		HIP_CHECK(ctx, hipMalloc(&degL[l], nelem * sizeof(uint32_t)));
		HIP_CHECK(ctx, hipMemset(degL[l], 0, nelem * sizeof(uint32_t)));
		h->dtrack[l] = std::make_unique<sdbv_hnsw::DirtyTrack>();
		h->dtrack[l]->flag.assign(nelem, 0);
		h->dtrack[l]->list.assign(nelem, 0);
		for (uint64_t i = 0; i < base; i++) { // pre-existing graph
			if (h->layers[l].has((uint32_t)i)) {
				h->mark_dirty(l, (uint32_t)i);
				has_members[l] = 1;
			}
		}
	}
	h->adj_dev = adjL[0];
	h->deg_dev = degL[0];
	h->adj_stride = strideL[0];
	h->adj_nodes = nelem;
	h->dev_rows = nelem;
	const uint64_t vwords = (nelem + 31) / 32;
	{
		uint64_t need = (uint64_t)chunk * vwords * sizeof(uint32_t);
		if (h->vis_cap < need) {
			if (h->vis_dev)
				(void)hipFree(h->vis_dev);
			h->vis_dev = nullptr;
			h->vis_cap = 0;
			HIP_CHECK(ctx, hipMalloc(&h->vis_dev, need));
			h->vis_cap = need;
		}
	}
	// ---- per-launch scratch (capacity = chunk, reused) ----
	float *Qd = nullptr;
	double *qnd = nullptr, *epdd = nullptr, *outd = nullptr;
	uint32_t *epsd = nullptr, *epoffd = nullptr;
	uint32_t *outr = nullptr, *outc = nullptr, *outf = nullptr;
	uint32_t *upd_ids_dev = nullptr, *upd_deg_dev = nullptr,
	         *upd_edges_dev = nullptr;
	uint32_t *listsd = nullptr, *loffd = nullptr, *seld = nullptr,
	         *selcntd = nullptr;
	uint64_t *moffd = nullptr;
	double *matsd = nullptr;
	uint64_t upd_cap = 0, lists_cap = 0, mats_cap = 0, sel_cap = 0,
	         loff_cap = 0;
	auto cleanup = [&] {
		adjL[0] = nullptr; // owned by h->adj_dev (hnsw_free_device_state)
		degL[0] = nullptr;
		for (uint32_t l = 1; l < ndev_layers; l++) {
			if (adjL[l])
				(void)hipFree(adjL[l]);
			if (degL[l])
				(void)hipFree(degL[l]);
		}
		for (void *p : {(void *)Qd, (void *)qnd, (void *)epdd, (void *)epsd,
		                (void *)epoffd, (void *)outr, (void *)outd,
		                (void *)outc, (void *)outf, (void *)upd_ids_dev,
		                (void *)upd_deg_dev, (void *)upd_edges_dev,
		                (void *)listsd, (void *)loffd, (void *)seld,
		                (void *)selcntd, (void *)moffd, (void *)matsd})
			if (p)
				(void)hipFree(p);
		h->dtrack.clear();
	};
#define BGPU_CHECK(call)                                                     \
	do {                                                                     \
		hipError_t err_ = (call);                                            \
		if (err_ != hipSuccess) {                                            \
			ctx->err = std::string("hip: ") + hipGetErrorString(err_);       \
			cleanup();                                                       \
			return SDBV_ERR_HIP;                                             \
		}                                                                    \
	} while (0)
	BGPU_CHECK(hipMalloc(&Qd, (uint64_t)chunk * d * sizeof(float)));
	BGPU_CHECK(hipMalloc(&qnd, chunk * sizeof(double)));
	BGPU_CHECK(hipMalloc(&epdd,
	                     (uint64_t)chunk * (efc + 1) * sizeof(double)));
	BGPU_CHECK(hipMalloc(&epsd,
	                     (uint64_t)chunk * (efc + 1) * sizeof(uint32_t)));
	BGPU_CHECK(hipMalloc(&epoffd, (chunk + 1) * sizeof(uint32_t)));
	BGPU_CHECK(hipMalloc(&outr, (uint64_t)chunk * efc * sizeof(uint32_t)));
	BGPU_CHECK(hipMalloc(&outd, (uint64_t)chunk * efc * sizeof(double)));
	BGPU_CHECK(hipMalloc(&outc, chunk * sizeof(uint32_t)));
	BGPU_CHECK(hipMalloc(&outf, chunk * sizeof(uint32_t)));
	// host staging
	std::vector<float> Qh;
	std::vector<double> qnh, epdh;
	std::vector<uint32_t> epsh, epoffh, outch, outfh, outrh;
	std::vector<double> outdh;
	std::vector<uint32_t> upd_ids, upd_deg, upd_edges;
	std::vector<uint32_t> listsh, loffh, selh, selcnth;
	std::vector<uint64_t> moffh;

	double t_epinit = 0, t_sync = 0, t_kern = 0, t_selA = 0, t_link = 0,
	       t_upperk = 0;
	auto now = [] { return std::chrono::steady_clock::now(); };
	auto secs = [](std::chrono::steady_clock::time_point a,
	               std::chrono::steady_clock::time_point b) {
		return std::chrono::duration<double>(b - a).count();
	};

	auto sync_adj = [&](uint32_t l) -> int {
		auto &D = *h->dtrack[l];
		upd_ids.clear();
		upd_deg.clear();
		uint32_t maxdeg = 0;
		auto &L = h->layers[l];
		uint32_t nd = D.n.load(std::memory_order_relaxed);
		for (uint32_t li = 0; li < nd; li++) {
			uint32_t i = D.list[li];
			uint32_t dg = (uint32_t)L.edges[i].size();
			maxdeg = std::max(maxdeg, dg);
			upd_ids.push_back(i);
			upd_deg.push_back(dg);
		}
		if (maxdeg > strideL[l]) {
			uint32_t ns = (maxdeg + 8 + 7) & ~7u;
			(void)hipFree(adjL[l]);
			adjL[l] = nullptr;
			BGPU_CHECK(hipMalloc(&adjL[l],
			                     nelem * (uint64_t)ns * sizeof(uint32_t)));
			strideL[l] = ns;
			if (l == 0) {
				h->adj_dev = adjL[0];
				h->adj_stride = ns;
			}
			upd_ids.clear();
			upd_deg.clear();
			for (uint64_t i = 0; i < nelem; i++)
				if (!L.edges[i].empty() || D.flag[i]) {
					upd_ids.push_back((uint32_t)i);
					upd_deg.push_back((uint32_t)L.edges[i].size());
				}
		}
		for (uint32_t li = 0; li < nd; li++)
			D.flag[D.list[li]] = 0;
		D.n.store(0, std::memory_order_relaxed);
		uint64_t cnt = upd_ids.size();
		if (cnt == 0)
			return SDBV_OK;
		const uint32_t stride = strideL[l];
		upd_edges.assign(cnt * stride, 0);
		for (uint64_t i = 0; i < cnt; i++) {
			const auto &e = L.edges[upd_ids[i]];
			std::copy(e.begin(), e.end(), upd_edges.begin() + i * stride);
		}
		if (upd_cap < (uint64_t)cnt * stride) {
			for (void **p : {(void **)&upd_ids_dev, (void **)&upd_deg_dev,
			                 (void **)&upd_edges_dev})
				if (*p) {
					(void)hipFree(*p);
					*p = nullptr;
				}
			uint64_t cap = ((uint64_t)cnt * stride * 3) / 2;
			BGPU_CHECK(hipMalloc(&upd_ids_dev, cap * sizeof(uint32_t)));
			BGPU_CHECK(hipMalloc(&upd_deg_dev, cap * sizeof(uint32_t)));
			BGPU_CHECK(hipMalloc(&upd_edges_dev, cap * sizeof(uint32_t)));
			upd_cap = cap;
		}
		BGPU_CHECK(hipMemcpyAsync(upd_ids_dev, upd_ids.data(),
		                          cnt * sizeof(uint32_t),
		                          hipMemcpyHostToDevice, ctx->stream));
		BGPU_CHECK(hipMemcpyAsync(upd_deg_dev, upd_deg.data(),
		                          cnt * sizeof(uint32_t),
		                          hipMemcpyHostToDevice, ctx->stream));
		BGPU_CHECK(hipMemcpyAsync(upd_edges_dev, upd_edges.data(),
		                          cnt * stride * sizeof(uint32_t),
		                          hipMemcpyHostToDevice, ctx->stream));
		uint64_t threads = cnt * stride;
		hipLaunchKernelGGL(k_adj_scatter,
		                   dim3((uint32_t)((threads + 255) / 256)), dim3(256),
		                   0, ctx->stream, upd_ids_dev, upd_deg_dev,
		                   upd_edges_dev, (uint32_t)cnt, stride, adjL[l],
		                   degL[l]);
		return SDBV_OK;
	};

	auto device_select = [&](const std::vector<uint32_t> &lists,
	                         const std::vector<uint32_t> &loff,
	                         uint32_t nlists, uint32_t m_max) -> int {
		moffh.resize(nlists + 1);
		uint64_t mo = 0;
		for (uint32_t i = 0; i < nlists; i++) {
			moffh[i] = mo;
			uint64_t len = loff[i + 1] - loff[i];
			mo += len * len;
		}
		moffh[nlists] = mo;
		if (lists_cap < lists.size()) {
			if (listsd)
				(void)hipFree(listsd);
			listsd = nullptr;
			BGPU_CHECK(hipMalloc(&listsd,
			                     lists.size() * 3 / 2 * sizeof(uint32_t)));
			lists_cap = lists.size() * 3 / 2;
		}
		if (loff_cap < nlists + 1) {
			for (void **p : {(void **)&loffd, (void **)&moffd,
			                 (void **)&selcntd})
				if (*p) {
					(void)hipFree(*p);
					*p = nullptr;
				}
			uint64_t cap = (nlists + 1) * 3 / 2;
			BGPU_CHECK(hipMalloc(&loffd, cap * sizeof(uint32_t)));
			BGPU_CHECK(hipMalloc(&moffd, cap * sizeof(uint64_t)));
			BGPU_CHECK(hipMalloc(&selcntd, cap * sizeof(uint32_t)));
			loff_cap = cap;
		}
		if (mats_cap < mo) {
			if (matsd)
				(void)hipFree(matsd);
			matsd = nullptr;
			BGPU_CHECK(hipMalloc(&matsd, mo * 5 / 4 * sizeof(double)));
			mats_cap = mo * 5 / 4;
		}
		if (sel_cap < (uint64_t)nlists * m_max) {
			if (seld)
				(void)hipFree(seld);
			seld = nullptr;
			BGPU_CHECK(hipMalloc(&seld, (uint64_t)nlists * m_max * 3 / 2 *
			                                sizeof(uint32_t)));
			sel_cap = (uint64_t)nlists * m_max * 3 / 2;
		}
		BGPU_CHECK(hipMemcpyAsync(listsd, lists.data(),
		                          lists.size() * sizeof(uint32_t),
		                          hipMemcpyHostToDevice, ctx->stream));
		BGPU_CHECK(hipMemcpyAsync(loffd, loff.data(),
		                          (nlists + 1) * sizeof(uint32_t),
		                          hipMemcpyHostToDevice, ctx->stream));
		BGPU_CHECK(hipMemcpyAsync(moffd, moffh.data(),
		                          (nlists + 1) * sizeof(uint64_t),
		                          hipMemcpyHostToDevice, ctx->stream));
		hipLaunchKernelGGL(k_pair_mats, dim3(nlists), dim3(256), 0,
		                   ctx->stream, h->rm_dev, h->norms_dev, d,
		                   (int)h->metric, listsd, loffd, moffd, matsd);
		hipLaunchKernelGGL(k_heur_select, dim3(nlists), dim3(64), 0,
		                   ctx->stream, listsd, loffd, moffd, matsd, m_max,
		                   h->keep ? 1 : 0, seld, selcntd);
		selh.resize((uint64_t)nlists * m_max);
		selcnth.resize(nlists);
		BGPU_CHECK(hipMemcpyAsync(selh.data(), seld,
		                          (uint64_t)nlists * m_max *
		                              sizeof(uint32_t),
		                          hipMemcpyDeviceToHost, ctx->stream));
		BGPU_CHECK(hipMemcpyAsync(selcnth.data(), selcntd,
		                          nlists * sizeof(uint32_t),
		                          hipMemcpyDeviceToHost, ctx->stream));
		BGPU_CHECK(hipStreamSynchronize(ctx->stream));
		BGPU_CHECK(hipGetLastError());
		return SDBV_OK;
	};

	// launch the persistent-kernel search for a subset of chunk items at
	// layer l: queries/norms/eps packed from item indices; results land in
	// outr/outd/outc/outf at the SUBSET ordinal positions.
	std::vector<ApplyItem> items;
	std::vector<PQ> eps_of;
	std::vector<uint32_t> lvl_of;
	auto launch_search = [&](uint32_t l, const std::vector<uint32_t> &sub,
	                         uint32_t k, uint32_t ef) -> int {
		uint32_t b = (uint32_t)sub.size();
		Qh.resize((uint64_t)b * d);
		qnh.resize(b);
		epoffh.resize(b + 1);
		epsh.clear();
		epdh.clear();
		for (uint32_t j = 0; j < b; j++) {
			uint32_t q_id = items[sub[j]].q_id;
			std::memcpy(Qh.data() + (uint64_t)j * d, vec(h, q_id),
			            d * sizeof(float));
			qnh[j] = h->metric == SDBV_METRIC_COSINE ? h->norms[q_id] : 0;
			epoffh[j] = (uint32_t)epsh.size();
			for (auto &e : eps_of[sub[j]].to_vec()) {
				epsh.push_back(e.second);
				epdh.push_back(e.first);
			}
		}
		epoffh[b] = (uint32_t)epsh.size();
		BGPU_CHECK(hipMemcpyAsync(Qd, Qh.data(),
		                          (uint64_t)b * d * sizeof(float),
		                          hipMemcpyHostToDevice, ctx->stream));
		BGPU_CHECK(hipMemcpyAsync(qnd, qnh.data(), b * sizeof(double),
		                          hipMemcpyHostToDevice, ctx->stream));
		BGPU_CHECK(hipMemcpyAsync(epsd, epsh.data(),
		                          epsh.size() * sizeof(uint32_t),
		                          hipMemcpyHostToDevice, ctx->stream));
		BGPU_CHECK(hipMemcpyAsync(epdd, epdh.data(),
		                          epdh.size() * sizeof(double),
		                          hipMemcpyHostToDevice, ctx->stream));
		BGPU_CHECK(hipMemcpyAsync(epoffd, epoffh.data(),
		                          (b + 1) * sizeof(uint32_t),
		                          hipMemcpyHostToDevice, ctx->stream));
		BGPU_CHECK(hipMemsetAsync(h->vis_dev, 0,
		                          (uint64_t)b * vwords * sizeof(uint32_t),
		                          ctx->stream));
		hipLaunchKernelGGL(k_hnsw_search<1>, dim3(b), dim3(64), 0,
		                   ctx->stream, h->rm_dev, h->norms_dev, d,
		                   (int)h->metric, degL[l], adjL[l], strideL[l], Qd,
		                   qnd, epsd, epdd, epoffd, h->vis_dev, vwords, k,
		                   ef, outr, outd, outc, outf);
		outch.resize(b);
		outfh.resize(b);
		outrh.resize((uint64_t)b * k);
		outdh.resize((uint64_t)b * k);
		BGPU_CHECK(hipMemcpyAsync(outch.data(), outc, b * sizeof(uint32_t),
		                          hipMemcpyDeviceToHost, ctx->stream));
		BGPU_CHECK(hipMemcpyAsync(outfh.data(), outf, b * sizeof(uint32_t),
		                          hipMemcpyDeviceToHost, ctx->stream));
		BGPU_CHECK(hipMemcpyAsync(outrh.data(), outr,
		                          (uint64_t)b * k * sizeof(uint32_t),
		                          hipMemcpyDeviceToHost, ctx->stream));
		BGPU_CHECK(hipMemcpyAsync(outdh.data(), outd,
		                          (uint64_t)b * k * sizeof(double),
		                          hipMemcpyDeviceToHost, ctx->stream));
		BGPU_CHECK(hipStreamSynchronize(ctx->stream));
		BGPU_CHECK(hipGetLastError());
		return SDBV_OK;
	};

	// device select + phase B/C for the insert set at layer l. The w
	// windows were just searched into outrh/outdh (subset order).
	auto apply_layer = [&](uint32_t l, const std::vector<uint32_t> &ins)
	    -> int {
		uint32_t b = (uint32_t)ins.size();
		uint32_t m_max = h->layers[l].m_max;
		listsh.clear();
		loffh.assign(b + 1, 0);
		for (uint32_t j = 0; j < b; j++) {
			loffh[j] = (uint32_t)listsh.size();
			listsh.push_back(items[ins[j]].q_id);
			if (!(outfh[j] & HQ_FLAG_OVERFLOW))
				for (uint32_t i = 0; i < outch[j]; i++)
					listsh.push_back(outrh[(uint64_t)j * efc + i]);
		}
		loffh[b] = (uint32_t)listsh.size();
		int rc = device_select(listsh, loffh, b, m_max);
		if (rc)
			return rc;
		for (uint32_t j = 0; j < b; j++) {
			ApplyItem &it = items[ins[j]];
			if (outfh[j] & HQ_FLAG_OVERFLOW) {
				// exact host fallback (rare)
				PQ w;
				snapshot_search_layer_eps(h, l, it.q_id, eps_of[ins[j]],
				                          efc, w);
				eps_of[ins[j]] = w;
				const float *q_pt = vec(h, it.q_id);
				double qn = h->metric == SDBV_METRIC_COSINE
				                ? h->norms[it.q_id]
				                : 0;
				it.neighbors.clear();
				select_neighbors(h, h->layers[l], it.q_id, q_pt, qn,
				                 std::move(w), it.neighbors, false);
				continue;
			}
			it.neighbors.assign(
			    selh.begin() + (uint64_t)j * m_max,
			    selh.begin() + (uint64_t)j * m_max + selcnth[j]);
		}
		std::vector<ApplyItem *> ptrs;
		ptrs.reserve(b);
		for (uint32_t j = 0; j < b; j++)
			ptrs.push_back(&items[ins[j]]);
		auto tb0 = now();
		auto overfull = batched_apply_phaseB(h, l, ptrs);
		t_link += secs(tb0, now());
		if (!overfull.empty()) {
			listsh.clear();
			loffh.clear();
			std::vector<uint32_t> host_prunes, dev_nodes;
			for (uint32_t e : overfull) {
				const auto &conn = h->layers[l].edges[e];
				if (1 + conn.size() > HSEL_CAP) {
					host_prunes.push_back(e);
					continue;
				}
				dev_nodes.push_back(e);
				loffh.push_back((uint32_t)listsh.size());
				listsh.push_back(e);
				for (uint32_t nid : conn) {
					if (nid < h->elem_present.size() &&
					    !h->elem_present[nid])
						continue;
					listsh.push_back(nid);
				}
			}
			loffh.push_back((uint32_t)listsh.size());
			if (!dev_nodes.empty()) {
				rc = device_select(listsh, loffh,
				                   (uint32_t)dev_nodes.size(), m_max);
				if (rc)
					return rc;
				for (uint32_t li = 0; li < dev_nodes.size(); li++) {
					uint32_t e = dev_nodes[li];
					auto &ee = h->layers[l].edges[e];
					ee.assign(selh.begin() + (uint64_t)li * m_max,
					          selh.begin() + (uint64_t)li * m_max +
					              selcnth[li]);
					h->mark_dirty(l, e);
				}
			}
			if (!host_prunes.empty())
				batched_apply_phaseC_host(h, l, host_prunes, nthreads);
		}
		return SDBV_OK;
	};

	std::vector<std::pair<uint32_t, uint32_t>> upper2; // (q_id, level)
	// warm-up: see the host twin — the first elements of an empty graph
	// insert classically so batching starts on a connected graph
	const uint64_t warm_until = h->enter_point < 0 ? base + 64 : 0;
	for (uint64_t c0 = 0; c0 < n; c0 += chunk) {
		uint64_t c1 = std::min(n, c0 + chunk);
		auto tp0 = now();
		items.clear();
		eps_of.clear();
		lvl_of.clear();
		upper2.clear();
		for (uint64_t i = c0; i < c1; i++) {
			if (h->enter_point < 0 || base + i < warm_until) {
				insert_at(h, (uint32_t)(base + i), levels[i], false);
				for (uint32_t l = 0;
				     l < ndev_layers && l <= levels[i]; l++) {
					h->mark_dirty(l, (uint32_t)(base + i));
					has_members[l] = 1;
				}
				continue;
			}
			if (levels[i] >= 2) {
				// levels >= 2 (0.4% of elements): FULL classic host
				// inserts — progressive at every layer (half-inserted
				// elements must never be search-visible; their classic
				// applies mark the device adjacency via
				// layer_insert_apply's dirty hooks)
				upper2.push_back({(uint32_t)(base + i), levels[i]});
				continue;
			}
			items.push_back(ApplyItem{(uint32_t)(base + i), PQ{}, {}});
			eps_of.emplace_back();
			lvl_of.push_back(levels[i]);
		}
		if (!upper2.empty()) {
			std::atomic<uint64_t> cursor{0};
			auto worker = [&]() {
				uint64_t j;
				while ((j = cursor.fetch_add(1)) < upper2.size())
					insert_at(h, upper2[j].first, upper2[j].second, true);
			};
			std::vector<std::thread> ts;
			int nt = std::max(1,
			                  std::min<int>(nthreads, (int)upper2.size()));
			for (int t = 1; t < nt; t++)
				ts.emplace_back(worker);
			worker();
			for (auto &t : ts)
				t.join();
			for (uint32_t l = 0; l < ndev_layers; l++)
				has_members[l] = 1; // classic inserts joined every layer
		}
		if (items.empty())
			continue;
		// levels <= 1: host greedy descents through layers top..2
		{
			std::atomic<uint64_t> cursor{0};
			auto worker = [&]() {
				uint64_t j;
				while ((j = cursor.fetch_add(1)) < items.size()) {
					uint32_t q_id = items[j].q_id;
					double qn = h->metric == SDBV_METRIC_COSINE
					                ? h->norms[q_id]
					                : 0;
					uint32_t ep;
					double epd;
					descend_to_layer2(h, q_id, qn, &ep, &epd);
					eps_of[j].push(epd, ep);
				}
			};
			std::vector<std::thread> ts;
			int nt = std::max(1,
			                  std::min<int>(nthreads, (int)items.size()));
			for (int t = 1; t < nt; t++)
				ts.emplace_back(worker);
			worker();
			for (auto &t : ts)
				t.join();
		}
		auto tp1 = now();
		t_epinit += secs(tp0, tp1);
		// layer 1 (device): ef=1 descents for level-0 elements, efc
		// searches + batched apply for level >= 1
		if (nlayers >= 2) {
			const uint32_t l = 1;
			std::vector<uint32_t> desc, ins;
			for (uint32_t j = 0; j < (uint32_t)items.size(); j++)
				(lvl_of[j] >= l ? ins : desc).push_back(j);
			auto tu0 = now();
			int rc = sync_adj(l);
			if (rc)
				return rc;
			t_sync += secs(tu0, now());
			if (has_members[l] && !desc.empty()) {
				auto tk = now();
				rc = launch_search(l, desc, 1, 1);
				if (rc)
					return rc;
				t_upperk += secs(tk, now());
				for (uint32_t j = 0; j < desc.size(); j++) {
					if (outch[j] == 0)
						continue; // keep previous eps
					PQ ne2;
					ne2.push(outdh[(uint64_t)j * 1], outrh[j]);
					eps_of[desc[j]] = std::move(ne2);
				}
			}
			if (!ins.empty()) {
				if (has_members[l]) {
					auto tk = now();
					rc = launch_search(l, ins, efc, efc);
					if (rc)
						return rc;
					t_upperk += secs(tk, now());
					// cascade eps = w
					for (uint32_t j = 0; j < ins.size(); j++) {
						if (outfh[j] & HQ_FLAG_OVERFLOW)
							continue; // handled in apply_layer
						PQ ne2;
						for (uint32_t i = 0; i < outch[j]; i++)
							ne2.push(outdh[(uint64_t)j * efc + i],
							         outrh[(uint64_t)j * efc + i]);
						eps_of[ins[j]] = std::move(ne2);
					}
				} else {
					// empty layer: search degenerates to w = eps
					outch.assign(ins.size(), 0);
					outfh.assign(ins.size(), 0);
					outrh.resize(ins.size() * (uint64_t)efc);
					outdh.resize(ins.size() * (uint64_t)efc);
					for (uint32_t j = 0; j < ins.size(); j++) {
						auto v = eps_of[ins[j]].to_vec();
						outch[j] = (uint32_t)std::min<size_t>(v.size(),
						                                      efc);
						for (uint32_t i = 0; i < outch[j]; i++) {
							outrh[(uint64_t)j * efc + i] = v[i].second;
							outdh[(uint64_t)j * efc + i] = v[i].first;
						}
					}
				}
				auto ta = now();
				rc = apply_layer(l, ins);
				if (rc)
					return rc;
				t_selA += secs(ta, now());
				has_members[l] = 1;
			}
		}
		// layer 0: every item
		{
			std::vector<uint32_t> all(items.size());
			for (uint32_t j = 0; j < (uint32_t)items.size(); j++)
				all[j] = j;
			auto tu0 = now();
			int rc = sync_adj(0);
			if (rc)
				return rc;
			t_sync += secs(tu0, now());
			auto tk = now();
			if (has_members[0]) {
				rc = launch_search(0, all, efc, efc);
				if (rc)
					return rc;
			} else {
				outch.assign(all.size(), 0);
				outfh.assign(all.size(), 0);
				outrh.resize(all.size() * (uint64_t)efc);
				outdh.resize(all.size() * (uint64_t)efc);
				for (uint32_t j = 0; j < all.size(); j++) {
					auto v = eps_of[j].to_vec();
					outch[j] = (uint32_t)std::min<size_t>(v.size(), efc);
					for (uint32_t i = 0; i < outch[j]; i++) {
						outrh[(uint64_t)j * efc + i] = v[i].second;
						outdh[(uint64_t)j * efc + i] = v[i].first;
					}
				}
			}
			t_kern += secs(tk, now());
			auto ta = now();
			rc = apply_layer(0, all);
			if (rc)
				return rc;
			t_selA += secs(ta, now());
			has_members[0] = 1;
		}
		if (((c0 / chunk) & 63) == 63)
			fprintf(stderr,
			        "[sdbv build_gpu3] %llu/%llu epinit=%.1f sync=%.1f "
			        "upperk=%.1f kern0=%.1f apply=%.1f link=%.1f\n",
			        (unsigned long long)c1, (unsigned long long)n,
			        t_epinit, t_sync, t_upperk, t_kern, t_selA, t_link);
	}
#undef BGPU_CHECK
	hnsw_promote_ep(h);
	fprintf(stderr,
	        "[sdbv build_gpu3] n=%llu chunk=%u phases: epinit=%.1fs "
	        "sync=%.1fs upper-kernels=%.1fs kernel0=%.1fs "
	        "apply(selA+selC)=%.1fs link=%.1fs\n",
	        (unsigned long long)n, chunk, t_epinit, t_sync, t_upperk,
	        t_kern, t_selA, t_link);
	cleanup();
	h->dirty = true;
	return SDBV_OK;
}


uint64_t sdbv_hnsw_n(sdbv_hnsw *h) { return h ? h->next_id : 0; }
uint32_t sdbv_hnsw_layers(sdbv_hnsw *h) {
	return h ? (uint32_t)h->layers.size() : 0;
}
uint64_t sdbv_hnsw_l0_edge_count(sdbv_hnsw *h) {
	uint64_t c = 0;
	for (auto &e : h->layers[0].edges)
		c += e.size();
	return c;
}
void sdbv_hnsw_l0_export(sdbv_hnsw *h, uint32_t *offsets, uint32_t *edges) {
	uint32_t off = 0;
	for (uint64_t i = 0; i < h->next_id; i++) {
		offsets[i] = off;
		if (i < h->layers[0].edges.size())
			for (uint32_t e : h->layers[0].edges[i])
				edges[off++] = e;
	}
	offsets[h->next_id] = off;
}
/* Full per-layer export (CSR + membership) — lets the bench's cpu_baseline
 * leg import the exact product graph into the oracle searcher. */
uint64_t sdbv_hnsw_layer_edge_count(sdbv_hnsw *h, uint32_t l) {
	if (!h || l >= h->layers.size())
		return 0;
	uint64_t c = 0;
	for (auto &e : h->layers[l].edges)
		c += e.size();
	return c;
}
void sdbv_hnsw_layer_export(sdbv_hnsw *h, uint32_t l, uint32_t *offsets,
                            uint32_t *edges, uint8_t *in_layer) {
	uint32_t off = 0;
	const auto &L = h->layers[l];
	for (uint64_t i = 0; i < h->next_id; i++) {
		offsets[i] = off;
		in_layer[i] = L.has((uint32_t)i) ? 1 : 0;
		if (i < L.edges.size())
			for (uint32_t e : L.edges[i])
				edges[off++] = e;
	}
	offsets[h->next_id] = off;
}
int64_t sdbv_hnsw_enter_point(sdbv_hnsw *h) {
	return h ? h->enter_point : -1;
}
/* Zero-copy view of the host row-major vector store (n x d f32, element id
 * = row): the oracle import reads rows from here instead of a second copy. */
const float *sdbv_hnsw_vecs_ptr(sdbv_hnsw *h) {
	return h ? h->vecs.data() : nullptr;
}

// Frees the per-index device state so finalize can be called again after
// host-graph mutations (apply_pendings re-finalizes a dirty index).
static void hnsw_free_device_state(sdbv_hnsw *h) {
	for (void **p : {(void **)&h->rows_dev, (void **)&h->dout_dev,
	                 (void **)&h->q_dev, (void **)&h->rm_dev,
	                 (void **)&h->offsets_dev, (void **)&h->edges_dev,
	                 (void **)&h->norms_dev, (void **)&h->vis_dev,
	                 (void **)&h->adj_dev, (void **)&h->deg_dev}) {
		if (*p)
			(void)hipFree(*p);
		*p = nullptr;
	}
	if (h->rows_pinned) {
		(void)hipHostFree(h->rows_pinned);
		h->rows_pinned = nullptr;
	}
	if (h->dists_pinned) {
		(void)hipHostFree(h->dists_pinned);
		h->dists_pinned = nullptr;
	}
	h->vis_cap = 0;
	h->adj_stride = 0;
	h->adj_nodes = 0;
	h->dev_rows = 0;
	h->finalized = false;
}

int sdbv_hnsw_finalize(sdbv_hnsw *h, uint64_t table) {
	if (!h || !h->ctx || h->next_id == 0)
		return SDBV_ERR_BAD_ARG;
	hnsw_free_device_state(h);
	int rc = sdbv_stage_corpus(h->ctx, table, h->vecs.data(), nullptr,
	                           h->next_id, h->d, h->metric);
	if (rc)
		return rc;
	sdbv_ctx *ctx = h->ctx;
	// per-hop scratch sized from the ACTUAL max layer-0 degree: a
	// parallel/batched build can leave a node transiently above m0 (the
	// keep-back fold in layer_insert_apply), and a frontier copy larger
	// than the scratch would overflow it (round-1 advisor class of bug)
	h->max_deg = h->m0 + 1;
	for (uint64_t i = 0; i < h->next_id; i++)
		h->max_deg = std::max<uint32_t>(
		    h->max_deg, (uint32_t)h->layers[0].edges[i].size() + 1);
	HIP_CHECK(ctx, hipMalloc(&h->rows_dev, h->max_deg * sizeof(uint32_t)));
	HIP_CHECK(ctx, hipMalloc(&h->dout_dev, h->max_deg * sizeof(double)));
	HIP_CHECK(ctx, hipMalloc(&h->q_dev, h->d * sizeof(float)));
	HIP_CHECK(ctx, hipHostMalloc(&h->rows_pinned,
	                             h->max_deg * sizeof(uint32_t)));
	HIP_CHECK(ctx, hipHostMalloc(&h->dists_pinned,
	                             h->max_deg * sizeof(double)));
	// persistent-kernel graph state: row-major vectors + layer-0 CSR
	uint64_t n = h->next_id;
	HIP_CHECK(ctx, hipMalloc(&h->rm_dev, n * h->d * sizeof(float)));
	HIP_CHECK(ctx, hipMemcpy(h->rm_dev, h->vecs.data(),
	                         n * h->d * sizeof(float),
	                         hipMemcpyHostToDevice));
	{
		// Scrubbed CSR: dangling edges to removed elements are dropped at
		// export, so the device kernels (which have no elements map) see
		// exactly the live graph — equivalent to the reference's
		// get_vector None gate (layer.rs:206).
		std::vector<uint32_t> offsets(n + 1);
		std::vector<uint32_t> edges;
		uint64_t ec = 0;
		for (uint64_t i = 0; i < n; i++) {
			offsets[i] = (uint32_t)ec;
			for (uint32_t e : h->layers[0].edges[i])
				if (h->elem_present[e]) {
					edges.push_back(e);
					ec++;
				}
		}
		offsets[n] = (uint32_t)ec;
		HIP_CHECK(ctx, hipMalloc(&h->offsets_dev, (n + 1) * sizeof(uint32_t)));
		HIP_CHECK(ctx, hipMalloc(&h->edges_dev,
		                         std::max<uint64_t>(ec, 1) * sizeof(uint32_t)));
		HIP_CHECK(ctx, hipMemcpy(h->offsets_dev, offsets.data(),
		                         (n + 1) * sizeof(uint32_t),
		                         hipMemcpyHostToDevice));
		if (ec)
			HIP_CHECK(ctx, hipMemcpy(h->edges_dev, edges.data(),
			                         ec * sizeof(uint32_t),
			                         hipMemcpyHostToDevice));
	}
	if (h->metric == SDBV_METRIC_COSINE) {
		HIP_CHECK(ctx, hipMalloc(&h->norms_dev, n * sizeof(double)));
		HIP_CHECK(ctx, hipMemcpy(h->norms_dev, h->norms.data(),
		                         n * sizeof(double), hipMemcpyHostToDevice));
	}
	h->table = table;
	h->finalized = true;
	h->dirty = false;
	return SDBV_OK;
}

// Batched ef-search on the persistent kernel: one query per workgroup; host
// does the upper-layer descent (search_ep) per query, the kernel runs the
// full layer-0 best-first loop. Results are EXACTLY the reference's
// (to_vec_limit(k) then the builder's (dist total_cmp, id) order). Any query
// whose in-kernel candidate queue overflowed (HQ_FLAG_OVERFLOW, not observed
// on real workloads) is re-run on the exact per-hop path.
int sdbv_hnsw_knn_batch(sdbv_hnsw *h, const float *Q, uint32_t b, uint32_t k,
                        uint32_t ef, uint64_t *out_ids, double *out_dists,
                        uint32_t *out_ns) {
	using namespace hnsw;
	// k is bounded by the in-kernel w window (HQ_EF_CAP), not the scan's
	// MAX_K: the output extraction loops generically over k slots. k <= 64
	// is the GPU-validated regime; larger k (the snapshot build's efc
	// window, round 2) shares the same code path.
	if (!h || !h->finalized || b == 0 || k == 0 || k > HQ_EF_CAP ||
	    ef > HQ_EF_CAP)
		return SDBV_ERR_BAD_ARG;
	if (h->enter_point < 0) {
		for (uint32_t j = 0; j < b; j++)
			out_ns[j] = 0;
		return SDBV_OK;
	}
	sdbv_ctx *ctx = h->ctx;
	std::unique_lock<std::mutex> lk(ctx->mu);
	uint64_t n = h->next_id;

	// host: per-query norms (restated chain) + upper-layer descent
	std::vector<double> qnorms(b);
	std::vector<uint32_t> eps(b);
	std::vector<double> epd(b);
	for (uint32_t j = 0; j < b; j++) {
		const float *q = Q + (uint64_t)j * h->d;
		qnorms[j] = sqrt(host_sumsq_f32(q, h->d));
		uint32_t ep_id = (uint32_t)h->enter_point;
		double ep_dist = dist(h, q, qnorms[j], ep_id);
		for (size_t l = h->layers.size() - 1; l >= 1; l--) {
			PQ cand;
			cand.push(ep_dist, ep_id);
			std::unordered_set<uint32_t> visited{ep_id};
			PQ w = cand;
			search_layer_host(h, h->layers[l], q, qnorms[j], cand, visited, w,
			                  1, false);
			double dd;
			uint32_t ii;
			if (w.peek_first(&dd, &ii)) {
				ep_dist = dd;
				ep_id = ii;
			}
		}
		eps[j] = ep_id;
		epd[j] = ep_dist;
	}

	// device buffers
	uint64_t vwords = (n + 31) / 32;
	uint64_t vis_need = (uint64_t)b * vwords * sizeof(uint32_t);
	if (h->vis_cap < vis_need) {
		if (h->vis_dev)
			(void)hipFree(h->vis_dev);
		h->vis_dev = nullptr;
		h->vis_cap = 0;
		HIP_CHECK(ctx, hipMalloc(&h->vis_dev, vis_need));
		h->vis_cap = vis_need;
	}
	HIP_CHECK(ctx, hipMemsetAsync(h->vis_dev, 0, vis_need, ctx->stream));
	float *Qd = nullptr;
	double *qnd = nullptr, *epdd = nullptr, *outd = nullptr;
	uint32_t *epsd = nullptr, *outr = nullptr, *outc = nullptr, *outf = nullptr;
	HIP_CHECK(ctx, hipMalloc(&Qd, (uint64_t)b * h->d * sizeof(float)));
	HIP_CHECK(ctx, hipMalloc(&qnd, b * sizeof(double)));
	HIP_CHECK(ctx, hipMalloc(&epdd, b * sizeof(double)));
	HIP_CHECK(ctx, hipMalloc(&epsd, b * sizeof(uint32_t)));
	HIP_CHECK(ctx, hipMalloc(&outr, (uint64_t)b * k * sizeof(uint32_t)));
	HIP_CHECK(ctx, hipMalloc(&outd, (uint64_t)b * k * sizeof(double)));
	HIP_CHECK(ctx, hipMalloc(&outc, b * sizeof(uint32_t)));
	HIP_CHECK(ctx, hipMalloc(&outf, b * sizeof(uint32_t)));
	auto cleanup = [&] {
		for (void *p : {(void *)Qd, (void *)qnd, (void *)epdd, (void *)epsd,
		                (void *)outr, (void *)outd, (void *)outc,
		                (void *)outf})
			if (p)
				(void)hipFree(p);
	};
	(void)hipMemcpyAsync(Qd, Q, (uint64_t)b * h->d * sizeof(float),
	               hipMemcpyHostToDevice, ctx->stream);
	(void)hipMemcpyAsync(qnd, qnorms.data(), b * sizeof(double),
	               hipMemcpyHostToDevice, ctx->stream);
	(void)hipMemcpyAsync(epdd, epd.data(), b * sizeof(double),
	               hipMemcpyHostToDevice, ctx->stream);
	(void)hipMemcpyAsync(epsd, eps.data(), b * sizeof(uint32_t),
	               hipMemcpyHostToDevice, ctx->stream);

	auto t0 = std::chrono::steady_clock::now();
	hipLaunchKernelGGL(k_hnsw_search<0>, dim3(b), dim3(64), 0, ctx->stream,
	                   h->rm_dev, h->norms_dev, h->d, (int)h->metric,
	                   h->offsets_dev, h->edges_dev, 0u, Qd, qnd, epsd, epdd,
	                   (const uint32_t *)nullptr,
	                   h->vis_dev, vwords, k, ef, outr, outd, outc, outf);
	std::vector<uint32_t> h_rows((uint64_t)b * k), h_cnt(b), h_flags(b);
	std::vector<double> h_d((uint64_t)b * k);
	(void)hipMemcpyAsync(h_rows.data(), outr, h_rows.size() * sizeof(uint32_t),
	               hipMemcpyDeviceToHost, ctx->stream);
	(void)hipMemcpyAsync(h_d.data(), outd, h_d.size() * sizeof(double),
	               hipMemcpyDeviceToHost, ctx->stream);
	(void)hipMemcpyAsync(h_cnt.data(), outc, b * sizeof(uint32_t),
	               hipMemcpyDeviceToHost, ctx->stream);
	(void)hipMemcpyAsync(h_flags.data(), outf, b * sizeof(uint32_t),
	               hipMemcpyDeviceToHost, ctx->stream);
	if (hipStreamSynchronize(ctx->stream) != hipSuccess ||
	    hipGetLastError() != hipSuccess) {
		ctx->err = "k_hnsw_search failed";
		cleanup();
		return SDBV_ERR_HIP;
	}
	ctx->stats.last_scan_kernel_ms =
	    std::chrono::duration<double, std::milli>(
	        std::chrono::steady_clock::now() - t0)
	        .count();
	ctx->stats.last_rows_scanned = n;
	cleanup();

	uint32_t overflowed = 0;
	for (uint32_t j = 0; j < b; j++)
		if (h_flags[j] & 1u)
			overflowed++;

	// final (dist total_cmp, id) order per query (knn.rs:363)
	for (uint32_t j = 0; j < b; j++) {
		if (h_flags[j] & 1u)
			continue; // re-run below on the exact per-hop path
		uint32_t m = h_cnt[j];
		std::vector<std::pair<std::pair<uint64_t, uint32_t>, double>> fin(m);
		for (uint32_t i = 0; i < m; i++) {
			double dd = h_d[(uint64_t)j * k + i];
			fin[i] = {{total_key(dd), h_rows[(uint64_t)j * k + i]}, dd};
		}
		std::sort(fin.begin(), fin.end());
		out_ns[j] = m;
		for (uint32_t i = 0; i < m; i++) {
			out_ids[(uint64_t)j * k + i] = fin[i].first.second;
			out_dists[(uint64_t)j * k + i] = fin[i].second;
		}
	}
	if (overflowed) {
		// exact fallback for overflowed queries (still the GPU gather path;
		// sdbv_hnsw_knn takes the ctx mutex itself)
		lk.unlock();
		for (uint32_t j = 0; j < b; j++)
			if (h_flags[j] & 1u) {
				int rc = sdbv_hnsw_knn(h, Q + (uint64_t)j * h->d, k, ef,
				                       out_ids + (uint64_t)j * k,
				                       out_dists + (uint64_t)j * k,
				                       &out_ns[j]);
				if (rc)
					return rc;
			}
	}
	return SDBV_OK;
}

// knn_search (hnsw/index.rs:270-335 without the host-kept parts): host
// upper-layer descent, then the layer-0 ef-search where each hop's
// neighbour distances come from the GPU gather kernel (the north_star's
// "batched gather + distance" design). Exact queue semantics preserved:
// distances are queue-independent, so batching them per hop does not change
// the reference's accept/update order.
int sdbv_hnsw_knn(sdbv_hnsw *h, const float *q, uint32_t k, uint32_t ef,
                  uint64_t *out_ids, double *out_dists, uint32_t *out_n) {
	using namespace hnsw;
	if (!h || !h->finalized)
		return SDBV_ERR_BAD_ARG;
	if (h->enter_point < 0) {
		*out_n = 0;
		return SDBV_OK;
	}
	sdbv_ctx *ctx = h->ctx;
	std::lock_guard<std::mutex> lk(ctx->mu);
	auto it = ctx->tables.find(h->table);
	if (it == ctx->tables.end())
		return SDBV_ERR_NO_TABLE;
	Table &t = it->second;

	double q_norm_d = 0;
	float q_sumsq = 0;
	{
		float p[8] = {0, 0, 0, 0, 0, 0, 0, 0};
		uint32_t i = 0;
		for (; i + 8 <= h->d; i += 8)
			for (uint32_t tt = 0; tt < 8; tt++)
				p[tt] += q[i + tt] * q[i + tt];
		q_sumsq = 0;
		q_sumsq += ((p[0] + p[4]) + (p[1] + p[5]));
		q_sumsq += ((p[2] + p[6]) + (p[3] + p[7]));
		for (; i < h->d; i++)
			q_sumsq += q[i] * q[i];
		q_norm_d = sqrt((double)q_sumsq);
	}

	// upper layers: greedy descent, host distances (tiny)
	uint32_t ep_id = (uint32_t)h->enter_point;
	double ep_dist = dist(h, q, q_norm_d, ep_id);
	for (size_t l = h->layers.size() - 1; l >= 1; l--) {
		PQ cand;
		cand.push(ep_dist, ep_id);
		std::unordered_set<uint32_t> visited{ep_id};
		PQ w = cand;
		search_layer_host(h, h->layers[l], q, q_norm_d, cand, visited, w, 1,
		                  false);
		double dd;
		uint32_t ii;
		if (w.peek_first(&dd, &ii)) {
			ep_dist = dd;
			ep_id = ii;
		}
	}

	// layer 0: ef-search with GPU batched neighbour expansion
	HIP_CHECK(ctx, hipMemcpyAsync(h->q_dev, q, h->d * sizeof(float),
	                              hipMemcpyHostToDevice, ctx->stream));
	const Layer &l0 = h->layers[0];
	PQ candidates, w;
	candidates.push(ep_dist, ep_id);
	w.push(ep_dist, ep_id);
	std::vector<bool> visited(h->next_id, false);
	visited[ep_id] = true;
	double fq = w.peek_last_dist(DBL_MAX);
	std::vector<uint32_t> frontier;
	std::vector<double> fdists(h->max_deg); // actual max degree
	double cd;
	uint32_t doc;
	double gpu_ms = 0;
	uint64_t gathered = 0;
	while (candidates.pop_first(&cd, &doc)) {
		if (cd > fq)
			break;
		frontier.clear();
		for (uint32_t e : l0.edges[doc])
			if (!visited[e]) {
				visited[e] = true;
				// elements.get_vector -> None (dangling edge to a
				// removed element): visited-marked then skipped
				if (!h->elem_present[e])
					continue;
				frontier.push_back(e);
			}
		if (frontier.empty())
			continue;
		auto hop_t0 = std::chrono::steady_clock::now();
		// ONE gather+distance launch for this hop's neighbours
		std::memcpy(h->rows_pinned, frontier.data(),
		            frontier.size() * sizeof(uint32_t));
		HIP_CHECK(ctx, hipMemcpyAsync(h->rows_dev, h->rows_pinned,
		                              frontier.size() * sizeof(uint32_t),
		                              hipMemcpyHostToDevice, ctx->stream));
		if (t.metric == SDBV_METRIC_COSINE)
			hipLaunchKernelGGL(k_gather_dist<0>, dim3((uint32_t)frontier.size()),
			                   dim3(64), 0, ctx->stream, t.cm, t.norms,
			                   t.n_pad, t.d, h->rows_dev,
			                   (uint32_t)frontier.size(), h->q_dev, q_norm_d,
			                   h->dout_dev);
		else
			hipLaunchKernelGGL(k_gather_dist<1>, dim3((uint32_t)frontier.size()),
			                   dim3(64), 0, ctx->stream, t.cm, t.norms,
			                   t.n_pad, t.d, h->rows_dev,
			                   (uint32_t)frontier.size(), h->q_dev, q_norm_d,
			                   h->dout_dev);
		HIP_CHECK(ctx, hipMemcpyAsync(h->dists_pinned, h->dout_dev,
		                              frontier.size() * sizeof(double),
		                              hipMemcpyDeviceToHost, ctx->stream));
		HIP_CHECK(ctx, hipStreamSynchronize(ctx->stream));
		std::memcpy(fdists.data(), h->dists_pinned,
		            frontier.size() * sizeof(double));
		gpu_ms += std::chrono::duration<double, std::milli>(
		              std::chrono::steady_clock::now() - hop_t0)
		              .count();
		gathered += frontier.size();
		// sequential accept/update in edge order (layer.rs:195-218)
		for (size_t i = 0; i < frontier.size(); i++) {
			double ed = fdists[i];
			uint32_t e = frontier[i];
			if (ed < fq || w.n < ef) {
				candidates.push(ed, e);
				w.push(ed, e);
				if (w.n > ef)
					w.pop_last();
				fq = w.peek_last_dist(DBL_MAX);
			}
		}
	}

	ctx->stats.last_scan_kernel_ms = gpu_ms; // gather sections (incl. copies)
	ctx->stats.last_merge_kernel_ms = 0;
	ctx->stats.last_rows_scanned = gathered;

	// to_vec_limit(k) in (dist, FIFO) order (knn.rs:92-104), then the
	// KnnResultBuilder (dist, id) final ordering (knn.rs:363)
	auto v = w.to_vec();
	size_t m = std::min<size_t>(k, v.size());
	std::vector<std::pair<std::pair<uint64_t, uint32_t>, double>> fin(m);
	for (size_t i = 0; i < m; i++)
		fin[i] = {{total_key(v[i].first), v[i].second}, v[i].first};
	std::sort(fin.begin(), fin.end());
	*out_n = (uint32_t)m;
	for (size_t i = 0; i < m; i++) {
		out_ids[i] = fin[i].first.second;
		out_dists[i] = fin[i].second;
	}
	return SDBV_OK;
}

int sdbv_hnsw_remove(sdbv_hnsw *h, uint64_t e_id) {
	if (!h || h->finalized)
		return SDBV_ERR_BAD_ARG;
	return hnsw::hnsw_remove(h, (uint32_t)e_id) ? 1 : 0;
}

// Host-side knn_search (same algorithm as the GPU per-hop path with host
// distances — the code path the build's efc-searches exercise). No device
// needed: lets CPU tests and quality audits search any host graph.
int sdbv_hnsw_knn_host(sdbv_hnsw *h, const float *q, uint32_t k, uint32_t ef,
                       uint64_t *out_ids, double *out_dists,
                       uint32_t *out_n) {
	using namespace hnsw;
	if (!h || !q || k == 0)
		return SDBV_ERR_BAD_ARG;
	if (h->enter_point < 0) {
		*out_n = 0;
		return SDBV_OK;
	}
	double q_norm = h->metric == SDBV_METRIC_COSINE
	                    ? sqrt(host_sumsq_f32(q, h->d))
	                    : 0;
	uint32_t ep_id = (uint32_t)h->enter_point;
	double ep_dist = dist(h, q, q_norm, ep_id);
	for (size_t l = h->layers.size() - 1; l >= 1; l--) {
		PQ cand;
		cand.push(ep_dist, ep_id);
		std::unordered_set<uint32_t> visited{ep_id};
		PQ w = cand;
		search_layer_host(h, h->layers[l], q, q_norm, cand, visited, w, 1,
		                  false);
		double dd;
		uint32_t ii;
		if (w.peek_first(&dd, &ii)) {
			ep_dist = dd;
			ep_id = ii;
		}
	}
	PQ cand, w;
	cand.push(ep_dist, ep_id);
	w.push(ep_dist, ep_id);
	std::unordered_set<uint32_t> visited{ep_id};
	search_layer_host(h, h->layers[0], q, q_norm, cand, visited, w, ef,
	                  false);
	auto v = w.to_vec();
	size_t m = std::min<size_t>(k, v.size());
	std::vector<std::pair<std::pair<uint64_t, uint32_t>, double>> fin(m);
	for (size_t i = 0; i < m; i++)
		fin[i] = {{total_key(v[i].first), v[i].second}, v[i].first};
	std::sort(fin.begin(), fin.end());
	*out_n = (uint32_t)m;
	for (size_t i = 0; i < m; i++) {
		out_ids[i] = fin[i].first.second;
		out_dists[i] = fin[i].second;
	}
	return SDBV_OK;
}

// Test hooks: drive the DoublePriorityQueue restatement directly so tests
// can replay the reference's own test_double_priority_queue sequence
// (knn.rs:735-790) against this implementation.
void *sdbv_test_pq_new() { return new hnsw::PQ(); }
void sdbv_test_pq_free(void *q) { delete (hnsw::PQ *)q; }
uint64_t sdbv_test_pq_len(void *q) { return ((hnsw::PQ *)q)->n; }
void sdbv_test_pq_push(void *q, double d, uint64_t id) {
	((hnsw::PQ *)q)->push(d, (uint32_t)id);
}
int sdbv_test_pq_peek_first(void *q, double *d, uint64_t *id) {
	uint32_t i;
	if (!((hnsw::PQ *)q)->peek_first(d, &i))
		return 0;
	*id = i;
	return 1;
}
int sdbv_test_pq_peek_last_dist(void *q, double *d) {
	auto *pq = (hnsw::PQ *)q;
	if (pq->n == 0)
		return 0;
	*d = pq->peek_last_dist(0);
	return 1;
}
int sdbv_test_pq_pop_first(void *q, double *d, uint64_t *id) {
	uint32_t i;
	if (!((hnsw::PQ *)q)->pop_first(d, &i))
		return 0;
	*id = i;
	return 1;
}
int sdbv_test_pq_pop_last(void *q, double *d, uint64_t *id) {
	auto *pq = (hnsw::PQ *)q;
	if (pq->n == 0)
		return 0;
	*d = pq->v.back().d;
	*id = pq->v.back().id;
	pq->pop_last();
	return 1;
}

} // extern "C"

// ===========================================================================
// Index layer — the operator surface of HnswIndex (hnsw/index.rs) with the
// parts the reference keeps behind the KV transaction held in host memory:
// the Hp pendings queue, the Hv vector->docs entries (VecDocs + Ids64), and
// the hi/hd record-key<->doc-id maps (HnswDocs). Record keys are opaque u64
// handles supplied by the host binding (INTEGRATION.md "record-key
// handles"). Searches run the pendings overlay on the host and the graph
// search on the GPU per-hop path once the index is finalized (re-finalizing
// automatically after writes); a host-only index (ctx == NULL) searches the
// host graph — that is the CPU-testable path, not a product fallback: a
// GPU-bound index always has a ctx.
// ===========================================================================

namespace vdocs {

// knn.rs:163-326 Ids64 — doc-id set with size-dependent representation.
// Vec1..Vec8 keep INSERTION order; the 9th insert collapses to Bits
// (iteration ascending); dropping back to exactly 8 keeps ascending order.
// insert/remove return "a new variant was produced" — the contract VecDocs
// persists on (Bits in-place mutations are dropped by the caller,
// docs.rs:374-382/:437-447; restated as-is).
struct Ids64 {
	std::vector<uint64_t> v;
	bool bits = false;
	size_t len() const { return v.size(); }
	bool empty() const { return v.empty(); }
	bool contains(uint64_t d) const {
		return std::find(v.begin(), v.end(), d) != v.end();
	}
	bool insert_ret_variant(uint64_t d) {
		if (contains(d))
			return false;
		if (!bits) {
			v.push_back(d);
			if (v.size() > 8) {
				std::sort(v.begin(), v.end());
				bits = true;
			}
			return true;
		}
		v.insert(std::lower_bound(v.begin(), v.end(), d), d);
		return false;
	}
	bool remove_ret_variant(uint64_t d, Ids64 *out) {
		if (bits) {
			auto it = std::lower_bound(v.begin(), v.end(), d);
			bool had = (it != v.end() && *it == d);
			if (had)
				v.erase(it);
			if (!had || v.size() != 8)
				return false;
			out->v = v;
			out->bits = false;
			return true;
		}
		switch (v.size()) {
		case 0:
			return false;
		case 1:
			if (v[0] == d) {
				out->v.clear();
				out->bits = false;
				return true;
			}
			return false;
		case 2:
			// knn.rs:266-268: first element != d becomes One — for a
			// non-member d this drops the second element (restated as-is)
			for (uint64_t x : v)
				if (x != d) {
					out->v = {x};
					out->bits = false;
					return true;
				}
			return false;
		default: {
			std::vector<uint64_t> f;
			for (uint64_t x : v)
				if (x != d)
					f.push_back(x);
			if (f.size() == v.size() - 1) {
				out->v = std::move(f);
				out->bits = false;
				return true;
			}
			return false;
		}
		}
	}
};

// knn.rs:363-437 KnnResultBuilder over VectorId (kind 0 = DocId < kind 1 =
// RecordKey, the enum's derived Ord).
struct Vid {
	uint8_t kind;
	uint64_t id;
	bool operator<(const Vid &o) const {
		if (kind != o.kind)
			return kind < o.kind;
		return id < o.id;
	}
};
struct Builder {
	size_t knn;
	struct Ent {
		uint64_t key;
		Vid vid;
		double dist;
		bool operator<(const Ent &o) const {
			if (key != o.key)
				return key < o.key;
			return vid < o.vid;
		}
	};
	std::set<Ent> pl;
	std::map<Vid, size_t> count;
	explicit Builder(size_t k) : knn(k) {}
	// knn.rs:386-394: plain f64 `>` against the current worst
	bool check_add(double submitted) const {
		if (pl.size() >= knn && !pl.empty() &&
		    submitted > std::prev(pl.end())->dist)
			return false;
		return true;
	}
	// knn.rs:410-433: returns true + *evicted when an id fell out of the
	// result entirely (count hit 0) — the filter-cache expiry signal.
	bool add_ret_evicted(double dist, Vid vid, Vid *evicted) {
		pl.insert(Ent{hnsw::total_key(dist), vid, dist});
		count[vid]++; // incremented even on duplicate set inserts
		if (pl.size() <= knn)
			return false;
		auto last = std::prev(pl.end());
		Vid ev = last->vid;
		pl.erase(last);
		auto it = count.find(ev);
		if (it != count.end()) {
			if (it->second <= 1) {
				count.erase(it);
				if (evicted)
					*evicted = ev;
				return true;
			}
			it->second--;
		}
		return false;
	}
	void add(double dist, Vid vid) { add_ret_evicted(dist, vid, nullptr); }
	void add_graph(double dist, const Ids64 &docs) {
		for (uint64_t doc : docs.v)
			add(dist, Vid{0, doc});
	}
};

} // namespace vdocs

struct sdbv_index {
	sdbv_hnsw *h;
	uint64_t table;
	struct ED {
		uint32_t e_id;
		vdocs::Ids64 docs;
	};
	// Hv entries: serialized vector bytes -> (element, docs)
	std::unordered_map<std::string, ED> vd;
	std::unordered_map<uint32_t, const std::string *> by_elem;
	// HnswDocs: hi/hd maps + recycled allocation (docs.rs:20-135)
	std::map<uint64_t, uint64_t> key2doc, doc2key;
	std::set<uint64_t> available;
	uint64_t next_doc_id = 0;
	// Hp pendings in appending order (index.rs:131-174)
	struct Pending {
		uint8_t kind; // 0 DocId, 1 RecordKey
		uint64_t id;
		std::vector<float> olds, news;
	};
	std::vector<Pending> pendings;
	// the reference's RwLock<HnswFlavor> (index.rs:55): one mutex here
	std::mutex mu;
};

namespace hnsw {
// layer.rs:320-338 are_all_docs_in_pending: an element with no VecDocs
// entry, or whose every doc is pending, counts as all-pending.
static bool idx_all_docs_pending(const IdxPend *p, uint32_t e_id) {
	if (!p->pending || p->pending->empty())
		return false;
	auto it = p->ix->by_elem.find(e_id);
	if (it != p->ix->by_elem.end()) {
		for (uint64_t doc : p->ix->vd.at(*it->second).docs.v)
			if (!p->pending->count(doc))
				return false;
	}
	return true;
}
} // namespace hnsw

// Distance of the query against a raw (not yet indexed) vector — the
// pendings overlay's Distance::calculate (idx/trees/vector.rs:660-672),
// same restated chain as the element path.
static double idx_dist_raw(const sdbv_hnsw *h, const float *q, double q_norm,
                           const float *v) {
	if (h->metric == SDBV_METRIC_COSINE) {
		double dot = hnsw::host_dot_f32(q, v, h->d);
		double v_norm = sqrt(hnsw::host_sumsq_f32(v, h->d));
		return 1.0 - dot / (q_norm * v_norm);
	}
	float acc = 0;
	for (uint32_t i = 0; i < h->d; i++) {
		float diff = q[i] - v[i];
		acc += diff * diff;
	}
	return sqrt((double)acc);
}

// docs.rs:64-90 resolve / :78-90 next_doc_id (smallest recycled id first).
static uint64_t idx_docs_resolve(sdbv_index *ix, uint64_t record_key) {
	auto it = ix->key2doc.find(record_key);
	if (it != ix->key2doc.end())
		return it->second;
	uint64_t doc_id;
	if (!ix->available.empty()) {
		doc_id = *ix->available.begin();
		ix->available.erase(ix->available.begin());
	} else {
		doc_id = ix->next_doc_id++;
	}
	ix->key2doc[record_key] = doc_id;
	ix->doc2key[doc_id] = record_key;
	return doc_id;
}

// docs.rs:113-135 HnswDocs::remove (recycle the id).
static void idx_docs_remove(sdbv_index *ix, uint64_t doc_id) {
	auto it = ix->doc2key.find(doc_id);
	if (it == ix->doc2key.end())
		return;
	ix->key2doc.erase(it->second);
	ix->doc2key.erase(it);
	ix->available.insert(doc_id);
}

// docs.rs:363-393 VecDocs::insert.
static int idx_vd_insert(sdbv_index *ix, const float *v, uint64_t doc_id) {
	std::string key((const char *)v, (size_t)ix->h->d * 4);
	auto it = ix->vd.find(key);
	if (it == ix->vd.end()) {
		uint32_t e_id = (uint32_t)ix->h->next_id;
		int rc = sdbv_hnsw_insert(ix->h, v);
		if (rc != SDBV_OK)
			return rc; // never silent: a dropped insert corrupts VecDocs
		auto r = ix->vd.emplace(std::move(key), sdbv_index::ED{e_id, {}});
		r.first->second.docs.v = {doc_id};
		ix->by_elem[e_id] = &r.first->first;
	} else {
		sdbv_index::ED ed = it->second; // owned copy, like tx.get
		if (ed.docs.insert_ret_variant(doc_id))
			it->second = ed; // persisted only on a new variant
	}
	return SDBV_OK;
}

// docs.rs:424-449 VecDocs::remove.
static void idx_vd_remove(sdbv_index *ix, const float *v, uint64_t doc_id) {
	std::string key((const char *)v, (size_t)ix->h->d * 4);
	auto it = ix->vd.find(key);
	if (it == ix->vd.end())
		return;
	sdbv_index::ED ed = it->second;
	vdocs::Ids64 new_docs;
	if (ed.docs.remove_ret_variant(doc_id, &new_docs)) {
		if (new_docs.empty()) {
			uint32_t e_id = ed.e_id;
			ix->by_elem.erase(e_id);
			ix->vd.erase(it);
			hnsw::hnsw_remove(ix->h, e_id);
		} else {
			ed.docs = new_docs;
			it->second = ed;
		}
	}
	// else: no new variant — the mutation is dropped (docs.rs:437-447)
}

// Host full graph search with pendings (knn_search, mod.rs:459-482 +
// search_ep :521-548): used for a host-only index (CPU tests).
static uint32_t idx_graph_search_host(sdbv_index *ix, const float *q,
                                      uint32_t k, uint32_t ef,
                                      const hnsw::IdxPend *pend,
                                      std::vector<std::pair<double, uint32_t>>
                                          &out) {
	using namespace hnsw;
	sdbv_hnsw *h = ix->h;
	if (h->enter_point < 0)
		return 0;
	double q_norm = h->metric == SDBV_METRIC_COSINE
	                    ? sqrt(host_sumsq_f32(q, h->d))
	                    : 0;
	uint32_t ep_id = (uint32_t)h->enter_point;
	double ep_dist = dist(h, q, q_norm, ep_id);
	for (size_t l = h->layers.size() - 1; l >= 1; l--) {
		PQ cand;
		cand.push(ep_dist, ep_id);
		std::unordered_set<uint32_t> visited{ep_id};
		PQ w = cand;
		search_layer_host(h, h->layers[l], q, q_norm, cand, visited, w, 1,
		                  false, pend);
		double dd;
		uint32_t ii;
		if (w.peek_first(&dd, &ii)) {
			ep_dist = dd;
			ep_id = ii;
		}
	}
	PQ cand;
	cand.push(ep_dist, ep_id);
	std::unordered_set<uint32_t> visited{ep_id};
	PQ w = cand;
	search_layer_host(h, h->layers[0], q, q_norm, cand, visited, w, ef,
	                  false, pend);
	auto v = w.to_vec(); // to_vec_limit(k), knn.rs:92-104
	uint32_t n = (uint32_t)std::min<size_t>(k, v.size());
	for (uint32_t i = 0; i < n; i++)
		out.push_back({v[i].first, v[i].second});
	return n;
}

// GPU per-hop graph search with pendings: the sdbv_hnsw_knn layer-0 loop
// (host queue + k_gather_dist batched expansion) with the pending gate on
// the candidates push. Caller holds ix->mu; takes ctx->mu itself.
static int idx_graph_search_gpu(sdbv_index *ix, const float *q, uint32_t k,
                                uint32_t ef, const hnsw::IdxPend *pend,
                                std::vector<std::pair<double, uint32_t>> &out,
                                uint32_t *out_n) {
	using namespace hnsw;
	sdbv_hnsw *h = ix->h;
	*out_n = 0;
	if (h->enter_point < 0)
		return SDBV_OK;
	sdbv_ctx *ctx = h->ctx;
	std::lock_guard<std::mutex> lk(ctx->mu);
	auto it = ctx->tables.find(h->table);
	if (it == ctx->tables.end())
		return SDBV_ERR_NO_TABLE;
	Table &t = it->second;
	double q_norm = h->metric == SDBV_METRIC_COSINE
	                    ? sqrt(host_sumsq_f32(q, h->d))
	                    : 0;
	// upper layers: host descent (tiny)
	uint32_t ep_id = (uint32_t)h->enter_point;
	double ep_dist = dist(h, q, q_norm, ep_id);
	for (size_t l = h->layers.size() - 1; l >= 1; l--) {
		PQ cand;
		cand.push(ep_dist, ep_id);
		std::unordered_set<uint32_t> visited{ep_id};
		PQ w = cand;
		search_layer_host(h, h->layers[l], q, q_norm, cand, visited, w, 1,
		                  false, pend);
		double dd;
		uint32_t ii;
		if (w.peek_first(&dd, &ii)) {
			ep_dist = dd;
			ep_id = ii;
		}
	}
	// layer 0: ef-search, one gather+distance launch per hop
	HIP_CHECK(ctx, hipMemcpyAsync(h->q_dev, q, h->d * sizeof(float),
	                              hipMemcpyHostToDevice, ctx->stream));
	const Layer &l0 = h->layers[0];
	PQ candidates, w;
	candidates.push(ep_dist, ep_id);
	w.push(ep_dist, ep_id);
	std::vector<bool> visited(h->next_id, false);
	visited[ep_id] = true;
	double fq = w.peek_last_dist(DBL_MAX);
	std::vector<uint32_t> frontier;
	std::vector<double> fdists(h->max_deg); // actual max degree
	double cd;
	uint32_t doc;
	while (candidates.pop_first(&cd, &doc)) {
		if (cd > fq)
			break;
		frontier.clear();
		for (uint32_t e : l0.edges[doc])
			if (!visited[e]) {
				visited[e] = true;
				if (!h->elem_present[e])
					continue;
				frontier.push_back(e);
			}
		if (frontier.empty())
			continue;
		std::memcpy(h->rows_pinned, frontier.data(),
		            frontier.size() * sizeof(uint32_t));
		HIP_CHECK(ctx, hipMemcpyAsync(h->rows_dev, h->rows_pinned,
		                              frontier.size() * sizeof(uint32_t),
		                              hipMemcpyHostToDevice, ctx->stream));
		if (t.metric == SDBV_METRIC_COSINE)
			hipLaunchKernelGGL(k_gather_dist<0>,
			                   dim3((uint32_t)frontier.size()), dim3(64), 0,
			                   ctx->stream, t.cm, t.norms, t.n_pad, t.d,
			                   h->rows_dev, (uint32_t)frontier.size(),
			                   h->q_dev, q_norm, h->dout_dev);
		else
			hipLaunchKernelGGL(k_gather_dist<1>,
			                   dim3((uint32_t)frontier.size()), dim3(64), 0,
			                   ctx->stream, t.cm, t.norms, t.n_pad, t.d,
			                   h->rows_dev, (uint32_t)frontier.size(),
			                   h->q_dev, q_norm, h->dout_dev);
		HIP_CHECK(ctx, hipMemcpyAsync(h->dists_pinned, h->dout_dev,
		                              frontier.size() * sizeof(double),
		                              hipMemcpyDeviceToHost, ctx->stream));
		HIP_CHECK(ctx, hipStreamSynchronize(ctx->stream));
		std::memcpy(fdists.data(), h->dists_pinned,
		            frontier.size() * sizeof(double));
		// sequential accept in edge order (layer.rs:195-218) with the
		// pending gate on candidates only (:209-212)
		for (size_t i = 0; i < frontier.size(); i++) {
			double ed = fdists[i];
			uint32_t e = frontier[i];
			if (ed < fq || w.n < ef) {
				if (!pend || !idx_all_docs_pending(pend, e))
					candidates.push(ed, e);
				w.push(ed, e);
				if (w.n > ef)
					w.pop_last();
				fq = w.peek_last_dist(DBL_MAX);
			}
		}
	}
	auto v = w.to_vec();
	uint32_t n = (uint32_t)std::min<size_t>(k, v.size());
	for (uint32_t i = 0; i < n; i++)
		out.push_back({v[i].first, v[i].second});
	*out_n = n;
	return SDBV_OK;
}

// ===========================================================================
// KV codec + cold-start staging pipeline: the
// reference's on-disk HNSW state — He element vectors (key/index/he.rs),
// Hn per-node edge lists (hn.rs), Hs graph state (hs.rs), Hv vector->docs
// entries (hv.rs) — parsed/emitted byte-exactly so a dumped surrealdb
// index bulk-loads straight into the host graph + device SoA with no
// re-insertion. Key layout (storekey, pinned by the reference's own key
// tests): `/*<ns u32 BE>*<db u32 BE>*<tb>\0+<ix u32 BE>!h<e|n|s|v>` +
// per-key fields (element u64 BE; layer u16 BE + node u64 BE; escaped
// revisioned vector). Embedded slices escape 0x00 -> 0x01 0x00 and
// 0x01 -> 0x01 0x01 with a 0x00 terminator (derived from the hv.rs
// golden bytes). Values use the `revision` crate 0.17.0 wire format
// (dependency pinned by Cargo.lock, source not vendored): varint revision
// tag + varint enum variant + varint lengths + fixed little-endian
// primitives — pinned by the hv.rs golden vectors at small sizes; beyond
// the single-byte range (values >= 251) the varints use bincode-style
// markers (0xFB + u16 LE, 0xFC + u32 LE, 0xFD + u64 LE), pinned by the
// catalog compat fixtures' 900/3600/86400 durations.
// ===========================================================================

namespace kvc {

// bincode-style unsigned varint (revision 0.17.0): < 251 one byte;
// 0xfb + u16 LE; 0xfc + u32 LE; 0xfd + u64 LE.
static void put_varint(std::vector<uint8_t> &b, uint64_t v) {
	if (v < 251) {
		b.push_back((uint8_t)v);
	} else if (v <= 0xFFFF) {
		b.push_back(0xFB);
		b.push_back((uint8_t)v);
		b.push_back((uint8_t)(v >> 8));
	} else if (v <= 0xFFFFFFFFull) {
		b.push_back(0xFC);
		for (int i = 0; i < 4; i++)
			b.push_back((uint8_t)(v >> (8 * i)));
	} else {
		b.push_back(0xFD);
		for (int i = 0; i < 8; i++)
			b.push_back((uint8_t)(v >> (8 * i)));
	}
}
static bool get_varint(const uint8_t *&p, const uint8_t *end, uint64_t *v) {
	if (p >= end)
		return false;
	uint8_t c = *p++;
	int extra;
	if (c < 251) {
		*v = c;
		return true;
	} else if (c == 0xFB) {
		extra = 2;
	} else if (c == 0xFC) {
		extra = 4;
	} else if (c == 0xFD) {
		extra = 8;
	} else {
		return false; // 0xFE (u128) unsupported
	}
	if (end - p < extra)
		return false;
	uint64_t out = 0;
	for (int i = 0; i < extra; i++)
		out |= (uint64_t)(*p++) << (8 * i);
	*v = out;
	return true;
}
template <typename T> static void put_le(std::vector<uint8_t> &b, T v) {
	uint8_t tmp[sizeof(T)];
	std::memcpy(tmp, &v, sizeof(T));
	b.insert(b.end(), tmp, tmp + sizeof(T));
}
template <typename T>
static bool get_le(const uint8_t *&p, const uint8_t *end, T *v) {
	if ((size_t)(end - p) < sizeof(T))
		return false;
	std::memcpy(v, p, sizeof(T));
	p += sizeof(T);
	return true;
}
template <typename T> static void put_be(std::vector<uint8_t> &b, T v) {
	for (int i = (int)sizeof(T) - 1; i >= 0; i--)
		b.push_back((uint8_t)(v >> (8 * i)));
}
template <typename T>
static bool get_be(const uint8_t *&p, const uint8_t *end, T *v) {
	if ((size_t)(end - p) < sizeof(T))
		return false;
	T out = 0;
	for (size_t i = 0; i < sizeof(T); i++)
		out = (out << 8) | *p++;
	*v = out;
	return true;
}

// storekey escaped-slice codec (embedded dynamic fields in keys)
static void put_escaped(std::vector<uint8_t> &b, const uint8_t *s, size_t n) {
	for (size_t i = 0; i < n; i++) {
		if (s[i] == 0x00 || s[i] == 0x01) {
			b.push_back(0x01);
			b.push_back(s[i]);
		} else {
			b.push_back(s[i]);
		}
	}
	b.push_back(0x00);
}
static bool get_escaped(const uint8_t *&p, const uint8_t *end,
                        std::vector<uint8_t> &out) {
	while (p < end) {
		uint8_t c = *p++;
		if (c == 0x00)
			return true; // terminator
		if (c == 0x01) {
			if (p >= end)
				return false;
			out.push_back(*p++);
		} else {
			out.push_back(c);
		}
	}
	return false;
}

// revisioned SerializedVector (idx/trees/vector.rs:33-41; F64=0 F32=1
// I64=2 I32=3 I16=4), F32 payloads only on the encode side (the HNSW
// default vector type, define.rs:1107).
static void enc_vec_f32(std::vector<uint8_t> &b, const float *v, uint32_t d) {
	put_varint(b, 1); // revision
	put_varint(b, 1); // variant F32
	put_varint(b, d);
	for (uint32_t i = 0; i < d; i++)
		put_le(b, v[i]);
}
static bool dec_vec_f32_cursor(const uint8_t *&p, const uint8_t *end,
                               std::vector<float> &out) {
	uint64_t rev, variant, len;
	if (!get_varint(p, end, &rev) || rev != 1)
		return false;
	if (!get_varint(p, end, &variant))
		return false;
	if (!get_varint(p, end, &len))
		return false;
	// bound before reserving: every element needs >= 4 payload bytes
	if (len > (uint64_t)(end - p) / 4 + 1)
		return false;
	out.clear();
	out.reserve(len);
	for (uint64_t i = 0; i < len; i++) {
		switch (variant) {
		case 1: { // F32
			float f;
			if (!get_le(p, end, &f))
				return false;
			out.push_back(f);
			break;
		}
		default:
			return false; // F64/I* corpora are not staged as f32 blindly
		}
	}
	return true;
}
static bool dec_vec_f32(const uint8_t *p, const uint8_t *end,
                        std::vector<float> &out) {
	return dec_vec_f32_cursor(p, end, out);
}

// graph.rs:104-113 node_to_val: u16 BE edge count + u64 BE edge ids.
static void enc_node(std::vector<uint8_t> &b,
                     const std::vector<uint32_t> &edges) {
	put_be(b, (uint16_t)edges.size());
	for (uint32_t e : edges)
		put_be(b, (uint64_t)e);
}
static bool dec_node(const uint8_t *p, const uint8_t *end,
                     std::vector<uint32_t> &out) {
	uint16_t n;
	if (!get_be(p, end, &n))
		return false;
	out.clear();
	out.reserve(n);
	for (uint16_t i = 0; i < n; i++) {
		uint64_t e;
		if (!get_be(p, end, &e))
			return false;
		out.push_back((uint32_t)e);
	}
	return true;
}

// revisioned HnswState (hnsw/mod.rs:61-71): Option<ElementId> enter_point
// (u8 tag 0/1 + u64 LE), next_element_id u64 LE, layer0 LayerState
// {version u64 LE, chunks u32 LE}, layers Vec<LayerState>.
struct HnswStateKV {
	bool has_ep = false;
	uint64_t enter_point = 0;
	uint64_t next_element_id = 0;
	uint64_t n_upper_layers = 0; // layers.len()
};
static void enc_state(std::vector<uint8_t> &b, const HnswStateKV &s) {
	put_varint(b, 1); // revision
	b.push_back(s.has_ep ? 1 : 0); // Option tag
	if (s.has_ep)
		put_varint(b, s.enter_point); // ElementId: unsigned -> varint
	put_varint(b, s.next_element_id);
	put_varint(b, 1);            // layer0 LayerState revision
	put_varint(b, 0);            // layer0.version (not tracked here)
	put_varint(b, 0);            // layer0.chunks == 0 (post-Hl format)
	put_varint(b, s.n_upper_layers);
	for (uint64_t i = 0; i < s.n_upper_layers; i++) {
		put_varint(b, 1); // LayerState revision
		put_varint(b, 0);
		put_varint(b, 0);
	}
}
static bool dec_state(const uint8_t *p, const uint8_t *end, HnswStateKV *s) {
	uint64_t rev;
	if (!get_varint(p, end, &rev) || rev != 1)
		return false;
	if (p >= end)
		return false;
	uint8_t tag = *p++;
	s->has_ep = tag != 0;
	if (s->has_ep && !get_varint(p, end, &s->enter_point))
		return false;
	if (!get_varint(p, end, &s->next_element_id))
		return false;
	uint64_t lrev, version, chunks;
	if (!get_varint(p, end, &lrev) || lrev != 1 ||
	    !get_varint(p, end, &version) || !get_varint(p, end, &chunks))
		return false; // layer0 state
	if (chunks != 0)
		return false; // legacy Hl chunks unsupported (post-migration only)
	if (!get_varint(p, end, &s->n_upper_layers))
		return false;
	for (uint64_t i = 0; i < s->n_upper_layers; i++)
		if (!get_varint(p, end, &lrev) || lrev != 1 ||
		    !get_varint(p, end, &version) ||
		    !get_varint(p, end, &chunks) || chunks != 0)
			return false;
	return true;
}

// revisioned ElementDocs (docs.rs:164-177): e_id varint + Ids64 (variant
// varint: Empty=0 One=1 Vec2=2..Vec8=8 Bits=9; doc ids as unsigned
// varints like every unsigned field). The reference's Bits variant is a
// serialized RoaringTreemap — not implemented here, so encode REFUSES a
// bits-mode set (callers return SDBV_ERR_UNSUPPORTED rather than emit a
// record a real surrealdb could not deserialize) and decode rejects
// variant 9.
static bool enc_element_docs(std::vector<uint8_t> &b, uint64_t e_id,
                             const vdocs::Ids64 &docs) {
	size_t n = docs.v.size();
	if (docs.bits || n > 8)
		return false; // Bits (roaring) — refuse, never emit malformed
	put_varint(b, 1);     // revision
	put_varint(b, e_id);  // ElementId: unsigned -> varint
	put_varint(b, 1);     // Ids64 revision
	put_varint(b, n);
	for (uint64_t d : docs.v)
		put_varint(b, d); // DocId: unsigned -> varint
	return true;
}
static bool dec_element_docs(const uint8_t *p, const uint8_t *end,
                             uint64_t *e_id, vdocs::Ids64 *docs) {
	uint64_t rev, variant;
	if (!get_varint(p, end, &rev) || rev != 1)
		return false;
	if (!get_varint(p, end, e_id))
		return false;
	if (!get_varint(p, end, &rev) || rev != 1)
		return false;
	if (!get_varint(p, end, &variant))
		return false;
	docs->v.clear();
	docs->bits = false;
	if (variant == 0)
		return true;
	if (variant > 8)
		return false; // Bits / roaring not supported in this revision
	for (uint64_t i = 0; i < variant; i++) {
		uint64_t d;
		if (!get_varint(p, end, &d))
			return false;
		docs->v.push_back(d);
	}
	return true;
}

// revisioned VectorPendingUpdate (hnsw/mod.rs:93-116): VectorId enum
// (DocId(u64 varint) = 0 | RecordKey(RecordIdKey) = 1; RecordIdKey
// revisioned enum — only Number(i64, fixed LE) maps onto the u64
// record-key handles of this boundary, other key kinds are the host's to
// apply) + old/new Vec<SerializedVector>.
struct PendingKV {
	uint8_t kind; // 0 DocId, 1 RecordKey(Number handle)
	uint64_t id;
	std::vector<float> olds, news; // n*d concatenated
};
static bool dec_vec_list(const uint8_t *&p, const uint8_t *end, uint32_t d,
                         std::vector<float> &out) {
	uint64_t n;
	if (!get_varint(p, end, &n))
		return false;
	out.clear();
	std::vector<float> one;
	for (uint64_t i = 0; i < n; i++) {
		if (!dec_vec_f32_cursor(p, end, one) || one.size() != d)
			return false;
		out.insert(out.end(), one.begin(), one.end());
	}
	return true;
}
static bool dec_pending(const uint8_t *p, const uint8_t *end, uint32_t d,
                        PendingKV *out) {
	uint64_t rev, variant;
	if (!get_varint(p, end, &rev) || rev != 1)
		return false; // VectorPendingUpdate revision
	if (!get_varint(p, end, &rev) || rev != 1)
		return false; // VectorId revision
	if (!get_varint(p, end, &variant))
		return false;
	if (variant == 0) { // DocId(u64)
		out->kind = 0;
		if (!get_varint(p, end, &out->id))
			return false;
	} else if (variant == 1) { // RecordKey(RecordIdKey)
		uint64_t krev, kvar;
		if (!get_varint(p, end, &krev) || krev != 1)
			return false;
		if (!get_varint(p, end, &kvar))
			return false;
		if (kvar != 0)
			return false; // only Number keys map to u64 handles here
		int64_t num;
		if (!get_le(p, end, &num))
			return false; // i64: signed -> fixed LE
		out->kind = 1;
		out->id = (uint64_t)num;
	} else {
		return false;
	}
	if (!dec_vec_list(p, end, d, out->olds))
		return false;
	if (!dec_vec_list(p, end, d, out->news))
		return false;
	return true;
}
static void enc_pending(std::vector<uint8_t> &b, uint32_t d,
                        const PendingKV &pk) {
	put_varint(b, 1); // VectorPendingUpdate revision
	put_varint(b, 1); // VectorId revision
	put_varint(b, pk.kind);
	if (pk.kind == 0) {
		put_varint(b, pk.id);
	} else {
		put_varint(b, 1);          // RecordIdKey revision
		put_varint(b, 0);          // Number variant
		put_le(b, (int64_t)pk.id); // signed fixed LE
	}
	put_varint(b, pk.olds.size() / d);
	for (size_t i = 0; i * d < pk.olds.size(); i++)
		enc_vec_f32(b, pk.olds.data() + i * d, d);
	put_varint(b, pk.news.size() / d);
	for (size_t i = 0; i * d < pk.news.size(); i++)
		enc_vec_f32(b, pk.news.data() + i * d, d);
}

// Key prefix `/*<ns>*<db>*<tb>\0+<ix>!h` + kind char.
static void key_prefix(std::vector<uint8_t> &b, uint32_t ns, uint32_t db,
                       const char *tb, uint32_t ix) {
	b.push_back('/');
	b.push_back('*');
	put_be(b, ns);
	b.push_back('*');
	put_be(b, db);
	b.push_back('*');
	b.insert(b.end(), tb, tb + strlen(tb));
	b.push_back(0);
	b.push_back('+');
	put_be(b, ix);
	b.push_back('!');
	b.push_back('h');
}
static void key_he(std::vector<uint8_t> &b, uint32_t ns, uint32_t db,
                   const char *tb, uint32_t ix, uint64_t element_id) {
	key_prefix(b, ns, db, tb, ix);
	b.push_back('e');
	put_be(b, element_id);
}
static void key_hn(std::vector<uint8_t> &b, uint32_t ns, uint32_t db,
                   const char *tb, uint32_t ix, uint16_t layer,
                   uint64_t node) {
	key_prefix(b, ns, db, tb, ix);
	b.push_back('n');
	put_be(b, layer);
	put_be(b, node);
}
static void key_hs(std::vector<uint8_t> &b, uint32_t ns, uint32_t db,
                   const char *tb, uint32_t ix) {
	key_prefix(b, ns, db, tb, ix);
	b.push_back('s');
}
static void key_hp(std::vector<uint8_t> &b, uint32_t ns, uint32_t db,
                   const char *tb, uint32_t ix, uint64_t appending_id) {
	key_prefix(b, ns, db, tb, ix);
	b.push_back('p');
	put_be(b, appending_id);
}
static void key_hv(std::vector<uint8_t> &b, uint32_t ns, uint32_t db,
                   const char *tb, uint32_t ix, const float *v, uint32_t d) {
	key_prefix(b, ns, db, tb, ix);
	b.push_back('v');
	std::vector<uint8_t> ser;
	enc_vec_f32(ser, v, d);
	put_escaped(b, ser.data(), ser.size());
}

// Parses a key: positions the kind char + the remainder. Returns 0 on a
// well-formed `!h?` key, else -1.
struct ParsedKey {
	char kind;
	const uint8_t *rest;
	const uint8_t *end;
};
static int parse_key(const uint8_t *k, size_t n, ParsedKey *out) {
	const uint8_t *p = k, *end = k + n;
	if (end - p < 12 || *p++ != '/' || *p++ != '*')
		return -1;
	p += 4; // ns
	if (p >= end || *p++ != '*')
		return -1;
	p += 4; // db
	if (p >= end || *p++ != '*')
		return -1;
	while (p < end && *p != 0)
		p++; // tb
	if (p >= end)
		return -1;
	p++; // NUL
	if (p >= end || *p++ != '+')
		return -1;
	p += 4; // ix
	if (end - p < 3 || *p++ != '!' || *p++ != 'h')
		return -1;
	out->kind = (char)*p++;
	out->rest = p;
	out->end = end;
	return 0;
}

} // namespace kvc

// ---------------------------------------------------------------------------
// Filtered KNN (hnsw/filter.rs + layer.rs:110-318): the WHERE-condition
// evaluation (is_record_truthy — KV fetch + expression compute) stays on
// the host side as a callback; the library keeps the FilterCache semantics
// (one evaluation per VectorId while cached, expired on builder eviction)
// and the accept/expand gating. Callback must be deterministic per call.
// ---------------------------------------------------------------------------

typedef int (*sdbv_truthy_cb)(void *user, uint8_t kind, uint64_t id);
typedef void (*sdbv_expire_cb)(void *user, uint8_t kind, uint64_t id);

namespace vdocs {

struct Filter {
	sdbv_truthy_cb cb;
	sdbv_expire_cb ex;
	void *user;
	std::map<std::pair<uint8_t, uint64_t>, bool> cache; // filter.rs:22
	bool truthy(uint8_t kind, uint64_t id) {
		auto key = std::make_pair(kind, id);
		auto it = cache.find(key);
		if (it != cache.end())
			return it->second;
		bool t = cb(user, kind, id) != 0;
		cache[key] = t;
		return t;
	}
	void expire(uint8_t kind, uint64_t id) { // filter.rs:141-144
		cache.erase({kind, id});
		if (ex)
			ex(user, kind, id);
	}
	bool any_doc_truthy(const Ids64 &docs) { // filter.rs:53-66
		for (uint64_t d : docs.v)
			if (truthy(0, d))
				return true;
		return false;
	}
};

} // namespace vdocs

// layer.rs:308-318 check_all_docs_in_pending (plain Ids64 variant).
static bool idx_check_all_docs_in_pending(const vdocs::Ids64 &docs,
                                          const std::set<uint64_t> *pending) {
	if (!pending || pending->empty())
		return false;
	for (uint64_t d : docs.v)
		if (!pending->count(d))
			return false;
	return true;
}

// layer.rs:278-306 add_if_truthy: docs looked up BY VECTOR.
static bool idx_add_if_truthy(sdbv_index *ix, uint32_t ef, hnsw::PQ &w,
                              const float *e_pt, double e_dist, uint32_t e_id,
                              vdocs::Filter &filter,
                              const std::set<uint64_t> *pending) {
	std::string key((const char *)e_pt, (size_t)ix->h->d * 4);
	auto it = ix->vd.find(key);
	if (it == ix->vd.end())
		return false;
	const vdocs::Ids64 &docs = it->second.docs;
	if (idx_check_all_docs_in_pending(docs, pending))
		return false;
	if (filter.any_doc_truthy(docs)) {
		w.push(e_dist, e_id);
		if (w.n > ef)
			w.pop_last();
		return true;
	}
	return false;
}

// layer.rs:226-275 search_with_filter, host distances (host-only index).
static void idx_search_l0_filter_host(sdbv_index *ix, const float *q,
                                      double q_norm, hnsw::PQ &candidates,
                                      std::vector<bool> &visited, hnsw::PQ &w,
                                      uint32_t ef, vdocs::Filter &filter,
                                      const std::set<uint64_t> *pending) {
	using namespace hnsw;
	sdbv_hnsw *h = ix->h;
	const Layer &l0 = h->layers[0];
	double f_dist = w.peek_last_dist(DBL_MAX);
	double cd;
	uint32_t doc;
	while (candidates.pop_first(&cd, &doc)) {
		if (cd > f_dist)
			break;
		for (uint32_t e : l0.edges[doc]) {
			if (visited[e])
				continue;
			visited[e] = true;
			if (!h->elem_present[e])
				continue;
			double ed = dist(h, q, q_norm, e);
			if (ed < f_dist || w.n < ef) {
				candidates.push(ed, e);
				if (idx_add_if_truthy(ix, ef, w, vec(h, e), ed, e, filter,
				                      pending))
					f_dist = w.peek_last_dist(DBL_MAX);
			}
		}
	}
}

// Same loop with the per-hop GPU gather for distances (finalized index).
// Caller holds ix->mu; takes ctx->mu itself.
static int idx_search_l0_filter_gpu(sdbv_index *ix, const float *q,
                                    double q_norm, hnsw::PQ &candidates,
                                    std::vector<bool> &visited, hnsw::PQ &w,
                                    uint32_t ef, vdocs::Filter &filter,
                                    const std::set<uint64_t> *pending) {
	using namespace hnsw;
	sdbv_hnsw *h = ix->h;
	sdbv_ctx *ctx = h->ctx;
	std::lock_guard<std::mutex> lk(ctx->mu);
	auto it = ctx->tables.find(h->table);
	if (it == ctx->tables.end())
		return SDBV_ERR_NO_TABLE;
	Table &t = it->second;
	HIP_CHECK(ctx, hipMemcpyAsync(h->q_dev, q, h->d * sizeof(float),
	                              hipMemcpyHostToDevice, ctx->stream));
	const Layer &l0 = h->layers[0];
	double f_dist = w.peek_last_dist(DBL_MAX);
	std::vector<uint32_t> frontier;
	std::vector<double> fdists(h->max_deg); // actual max degree
	double cd;
	uint32_t doc;
	while (candidates.pop_first(&cd, &doc)) {
		if (cd > f_dist)
			break;
		frontier.clear();
		for (uint32_t e : l0.edges[doc])
			if (!visited[e]) {
				visited[e] = true;
				if (!h->elem_present[e])
					continue;
				frontier.push_back(e);
			}
		if (frontier.empty())
			continue;
		std::memcpy(h->rows_pinned, frontier.data(),
		            frontier.size() * sizeof(uint32_t));
		HIP_CHECK(ctx, hipMemcpyAsync(h->rows_dev, h->rows_pinned,
		                              frontier.size() * sizeof(uint32_t),
		                              hipMemcpyHostToDevice, ctx->stream));
		if (t.metric == SDBV_METRIC_COSINE)
			hipLaunchKernelGGL(k_gather_dist<0>,
			                   dim3((uint32_t)frontier.size()), dim3(64), 0,
			                   ctx->stream, t.cm, t.norms, t.n_pad, t.d,
			                   h->rows_dev, (uint32_t)frontier.size(),
			                   h->q_dev, q_norm, h->dout_dev);
		else
			hipLaunchKernelGGL(k_gather_dist<1>,
			                   dim3((uint32_t)frontier.size()), dim3(64), 0,
			                   ctx->stream, t.cm, t.norms, t.n_pad, t.d,
			                   h->rows_dev, (uint32_t)frontier.size(),
			                   h->q_dev, q_norm, h->dout_dev);
		HIP_CHECK(ctx, hipMemcpyAsync(h->dists_pinned, h->dout_dev,
		                              frontier.size() * sizeof(double),
		                              hipMemcpyDeviceToHost, ctx->stream));
		HIP_CHECK(ctx, hipStreamSynchronize(ctx->stream));
		std::memcpy(fdists.data(), h->dists_pinned,
		            frontier.size() * sizeof(double));
		for (size_t i = 0; i < frontier.size(); i++) {
			double ed = fdists[i];
			uint32_t e = frontier[i];
			if (ed < f_dist || w.n < ef) {
				candidates.push(ed, e);
				if (idx_add_if_truthy(ix, ef, w, vec(h, e), ed, e, filter,
				                      pending))
					f_dist = w.peek_last_dist(DBL_MAX);
			}
		}
	}
	return SDBV_OK;
}

extern "C" {

int sdbv_index_create(sdbv_ctx *ctx, uint64_t table, uint32_t d,
                      uint8_t metric, uint32_t m, uint32_t m0, uint32_t efc,
                      int extend, int keep, uint64_t seed, double ml,
                      sdbv_index **out) {
	sdbv_hnsw *h = nullptr;
	int rc = sdbv_hnsw_create(ctx, d, metric, m, m0, efc, extend, keep, seed,
	                          ml, &h);
	if (rc)
		return rc;
	auto *ix = new sdbv_index();
	ix->h = h;
	ix->table = table;
	*out = ix;
	return SDBV_OK;
}

void sdbv_index_destroy(sdbv_index *ix) {
	if (!ix)
		return;
	sdbv_hnsw_destroy(ix->h);
	delete ix;
}

sdbv_hnsw *sdbv_index_hnsw(sdbv_index *ix) { return ix ? ix->h : nullptr; }

// Level-RNG state carry-over (an EXTENSION record): the reference draws
// insert levels from an entropy-seeded SmallRng (hnsw/mod.rs:263-266), so
// the sequence restarts arbitrarily on every process start and ANY state
// is reference-conformant. For this framework's determinism convention
// (same seed => same graph) a host may persist the state across cold
// starts and restore it after load.
uint64_t sdbv_index_level_rng(sdbv_index *ix) {
	return ix ? ix->h->rng_state : 0;
}
void sdbv_index_set_level_rng(sdbv_index *ix, uint64_t state) {
	if (ix)
		ix->h->rng_state = state;
}
uint64_t sdbv_index_doc_count(sdbv_index *ix) { return ix->doc2key.size(); }
uint64_t sdbv_index_pending_count(sdbv_index *ix) {
	return ix->pendings.size();
}

// HnswIndex::index (index.rs:138-186): resolve the id kind via the hi map
// and append one VectorPendingUpdate (old/new = n*d f32 each).
int sdbv_index_enqueue(sdbv_index *ix, uint64_t record_key, const float *olds,
                       uint32_t n_old, const float *news, uint32_t n_new) {
	if (!ix)
		return SDBV_ERR_BAD_ARG;
	std::lock_guard<std::mutex> lk(ix->mu);
	sdbv_index::Pending p;
	auto it = ix->key2doc.find(record_key);
	if (it != ix->key2doc.end()) {
		p.kind = 0;
		p.id = it->second;
	} else {
		p.kind = 1;
		p.id = record_key;
	}
	uint32_t d = ix->h->d;
	p.olds.assign(olds, olds + (size_t)n_old * d);
	p.news.assign(news, news + (size_t)n_new * d);
	ix->pendings.push_back(std::move(p));
	return SDBV_OK;
}

// index_pendings + index_pending (index.rs:188-257): drain in appending
// order; old-vector removals only for DocId pendings; empty new_vectors
// deletes the doc mapping; RecordKey ids resolve (allocate) at apply time.
int sdbv_index_apply_pendings(sdbv_index *ix, uint64_t *out_count) {
	if (!ix)
		return SDBV_ERR_BAD_ARG;
	std::lock_guard<std::mutex> lk(ix->mu);
	uint64_t count = 0;
	uint32_t d = ix->h->d;
	for (auto &p : ix->pendings) {
		if (p.kind == 0) {
			for (size_t i = 0; i * d < p.olds.size(); i++)
				idx_vd_remove(ix, p.olds.data() + i * d, p.id);
			if (p.news.empty())
				idx_docs_remove(ix, p.id);
		}
		if (!p.news.empty()) {
			uint64_t doc_id =
			    (p.kind == 0) ? p.id : idx_docs_resolve(ix, p.id);
			for (size_t i = 0; i * d < p.news.size(); i++) {
				int rc = idx_vd_insert(ix, p.news.data() + i * d, doc_id);
				if (rc != SDBV_OK)
					return rc;
			}
		}
		count++;
	}
	ix->pendings.clear();
	if (out_count)
		*out_count = count;
	return SDBV_OK;
}

// knn_search (index.rs:270-335 without record materialisation):
// search_pendings overlay + graph search (GPU per-hop once finalized,
// auto-refinalizing after writes) + VecDocs doc expansion through one
// KnnResultBuilder. Out arrays sized k; *out_n = entries returned. Entries
// ascending (dist total_cmp, VectorId); kind 0 = DocId, 1 = RecordKey.
int sdbv_index_knn(sdbv_index *ix, const float *q, uint32_t k, uint32_t ef,
                   uint8_t *out_kinds, uint64_t *out_ids, double *out_dists,
                   uint32_t *out_n) {
	if (!ix || !q || k == 0)
		return SDBV_ERR_BAD_ARG;
	std::lock_guard<std::mutex> lk(ix->mu);
	sdbv_hnsw *h = ix->h;
	vdocs::Builder builder(k);
	// search_pendings (index.rs:366-421)
	std::set<uint64_t> all_existing;
	std::map<vdocs::Vid, const std::vector<float> *> non_deleted;
	for (auto &p : ix->pendings) {
		if (p.kind == 0)
			all_existing.insert(p.id);
		vdocs::Vid vid{p.kind, p.id};
		if (p.news.empty())
			non_deleted.erase(vid);
		else
			non_deleted[vid] = &p.news;
	}
	if (!(all_existing.empty() && non_deleted.empty())) {
		double q_norm = h->metric == SDBV_METRIC_COSINE
		                    ? sqrt(hnsw::host_sumsq_f32(q, h->d))
		                    : 0;
		for (auto &e : non_deleted) {
			const std::vector<float> &vecs = *e.second;
			for (size_t i = 0; i * h->d < vecs.size(); i++) {
				double dd =
				    idx_dist_raw(h, q, q_norm, vecs.data() + i * h->d);
				if (builder.check_add(dd))
					builder.add(dd, e.first);
			}
		}
	}
	hnsw::IdxPend pend{&all_existing, ix};
	const hnsw::IdxPend *pp = all_existing.empty() ? nullptr : &pend;
	// graph search: GPU per-hop when a device context exists (finalizing
	// on first use / after writes), host path otherwise
	std::vector<std::pair<double, uint32_t>> neighbors;
	if (h->ctx && h->next_id > 0) {
		if (h->dirty || !h->finalized) {
			int rc = sdbv_hnsw_finalize(h, ix->table);
			if (rc)
				return rc;
		}
		uint32_t ng = 0;
		int rc = idx_graph_search_gpu(ix, q, k, ef, pp, neighbors, &ng);
		if (rc)
			return rc;
	} else {
		idx_graph_search_host(ix, q, k, ef, pp, neighbors);
	}
	// add_graph_results (index.rs:454-483)
	for (auto &nb : neighbors) {
		if (!builder.check_add(nb.first))
			continue;
		auto it = ix->by_elem.find(nb.second);
		if (it == ix->by_elem.end())
			continue; // get_vector/get_docs -> None
		builder.add_graph(nb.first, ix->vd.at(*it->second).docs);
	}
	uint32_t n = 0;
	for (const auto &e : builder.pl) {
		out_kinds[n] = e.vid.kind;
		out_ids[n] = e.vid.id;
		out_dists[n] = e.dist;
		n++;
	}
	*out_n = n;
	return SDBV_OK;
}

// knn_search with cond_filter (index.rs:270-335 + knn_search_with_filter
// mod.rs:484-515 + search_single_with_filter layer.rs:110-149): same flow
// as sdbv_index_knn but candidates expand unconditionally and w is gated
// by add_if_truthy (>=1 truthy doc, not all-pending). NOTE the reference
// seeds add_if_truthy with SEARCH.PT as the entry point's vector
// (layer.rs:125-135) — the entry point enters w only if the QUERY VECTOR
// itself is an indexed vector with a truthy doc; restated as-is.
int sdbv_index_knn_filtered(sdbv_index *ix, const float *q, uint32_t k,
                            uint32_t ef, sdbv_truthy_cb truthy,
                            sdbv_expire_cb expire, void *user,
                            uint8_t *out_kinds, uint64_t *out_ids,
                            double *out_dists, uint32_t *out_n) {
	using namespace hnsw;
	if (!ix || !q || k == 0 || !truthy)
		return SDBV_ERR_BAD_ARG;
	std::lock_guard<std::mutex> lk(ix->mu);
	sdbv_hnsw *h = ix->h;
	vdocs::Builder builder(k);
	vdocs::Filter filter{truthy, expire, user, {}};
	// search_pendings with the filter (index.rs:400-421)
	std::set<uint64_t> all_existing;
	std::map<vdocs::Vid, const std::vector<float> *> non_deleted;
	for (auto &p : ix->pendings) {
		if (p.kind == 0)
			all_existing.insert(p.id);
		vdocs::Vid vid{p.kind, p.id};
		if (p.news.empty())
			non_deleted.erase(vid);
		else
			non_deleted[vid] = &p.news;
	}
	if (!(all_existing.empty() && non_deleted.empty())) {
		double q_norm = h->metric == SDBV_METRIC_COSINE
		                    ? sqrt(host_sumsq_f32(q, h->d))
		                    : 0;
		for (auto &e : non_deleted) {
			if (!filter.truthy(e.first.kind, e.first.id))
				continue;
			const std::vector<float> &vecs = *e.second;
			for (size_t i = 0; i * h->d < vecs.size(); i++) {
				double dd =
				    idx_dist_raw(h, q, q_norm, vecs.data() + i * h->d);
				if (builder.check_add(dd)) {
					vdocs::Vid ev;
					if (builder.add_ret_evicted(dd, e.first, &ev))
						filter.expire(ev.kind, ev.id);
				}
			}
		}
	}
	const std::set<uint64_t> *pp =
	    all_existing.empty() ? nullptr : &all_existing;
	IdxPend pend{&all_existing, ix};
	const IdxPend *ep_pend = all_existing.empty() ? nullptr : &pend;
	// knn_search_with_filter (mod.rs:484-515)
	std::vector<std::pair<double, uint32_t>> neighbors;
	if (h->next_id > 0 && h->enter_point >= 0) {
		bool use_gpu = h->ctx != nullptr;
		if (use_gpu && (h->dirty || !h->finalized)) {
			int rc = sdbv_hnsw_finalize(h, ix->table);
			if (rc)
				return rc;
		}
		double q_norm = h->metric == SDBV_METRIC_COSINE
		                    ? sqrt(host_sumsq_f32(q, h->d))
		                    : 0;
		// search_ep: plain upper-layer descent with pending (mod.rs:520-548)
		uint32_t ep_id = (uint32_t)h->enter_point;
		double ep_dist = dist(h, q, q_norm, ep_id);
		for (size_t l = h->layers.size() - 1; l >= 1; l--) {
			PQ cand;
			cand.push(ep_dist, ep_id);
			std::unordered_set<uint32_t> visited{ep_id};
			PQ w = cand;
			search_layer_host(h, h->layers[l], q, q_norm, cand, visited, w,
			                  1, false, ep_pend);
			double dd;
			uint32_t ii;
			if (w.peek_first(&dd, &ii)) {
				ep_dist = dd;
				ep_id = ii;
			}
		}
		// search_single_with_filter (layer.rs:110-149; the search.pt seed)
		PQ candidates, w;
		candidates.push(ep_dist, ep_id);
		std::vector<bool> visited(h->next_id, false);
		visited[ep_id] = true;
		idx_add_if_truthy(ix, ef, w, q, ep_dist, ep_id, filter, pp);
		if (use_gpu) {
			int rc = idx_search_l0_filter_gpu(ix, q, q_norm, candidates,
			                                  visited, w, ef, filter, pp);
			if (rc)
				return rc;
		} else {
			idx_search_l0_filter_host(ix, q, q_norm, candidates, visited, w,
			                          ef, filter, pp);
		}
		auto v = w.to_vec();
		size_t m = std::min<size_t>(k, v.size());
		for (size_t i = 0; i < m; i++)
			neighbors.push_back({v[i].first, v[i].second});
	}
	// add_graph_results with eviction expiry (index.rs:353-359, :454-483)
	for (auto &nb : neighbors) {
		if (!builder.check_add(nb.first))
			continue;
		auto it = ix->by_elem.find(nb.second);
		if (it == ix->by_elem.end())
			continue;
		for (uint64_t docid : ix->vd.at(*it->second).docs.v) {
			vdocs::Vid ev;
			if (builder.add_ret_evicted(nb.first, vdocs::Vid{0, docid}, &ev))
				filter.expire(ev.kind, ev.id);
		}
	}
	uint32_t n = 0;
	for (const auto &e : builder.pl) {
		out_kinds[n] = e.vid.kind;
		out_ids[n] = e.vid.id;
		out_dists[n] = e.dist;
		n++;
	}
	*out_n = n;
	return SDBV_OK;
}

// ---- KV cold-start loader (He/Hn/Hs/Hv stream -> host graph + index) ----

struct sdbv_kvload {
	sdbv_ctx *ctx;
	uint32_t d;
	uint8_t metric;
	uint32_t m, m0, efc;
	int extend, keep;
	uint64_t seed;
	double ml;
	bool has_state = false;
	kvc::HnswStateKV state;
	std::map<uint64_t, std::vector<float>> elems;          // He
	std::map<std::pair<uint16_t, uint64_t>, std::vector<uint32_t>> nodes; // Hn
	struct HvEnt {
		std::vector<float> vec;
		uint64_t e_id;
		vdocs::Ids64 docs;
	};
	std::vector<HvEnt> hv; // Hv
	std::map<uint64_t, kvc::PendingKV> hp; // Hp by appending id
};

int sdbv_kvload_new(sdbv_ctx *ctx, uint32_t d, uint8_t metric, uint32_t m,
                    uint32_t m0, uint32_t efc, int extend, int keep,
                    uint64_t seed, double ml, sdbv_kvload **out) {
	if (d == 0 || (d % 4) != 0 || metric > SDBV_METRIC_EUCLIDEAN)
		return SDBV_ERR_BAD_ARG;
	auto *L = new sdbv_kvload();
	L->ctx = ctx;
	L->d = d;
	L->metric = metric;
	L->m = m;
	L->m0 = m0;
	L->efc = efc;
	L->extend = extend;
	L->keep = keep;
	L->seed = seed;
	L->ml = ml;
	*out = L;
	return SDBV_OK;
}

void sdbv_kvload_abort(sdbv_kvload *L) { delete L; }

// Feed one (key, value) pair from the index's KV range scan. Unknown `!h?`
// kinds (hd/hi/hp/hh records — host-kept) are skipped, not errors.
int sdbv_kvload_feed(sdbv_kvload *L, const uint8_t *key, uint64_t klen,
                     const uint8_t *val, uint64_t vlen) {
	if (!L || !key)
		return SDBV_ERR_BAD_ARG;
	kvc::ParsedKey pk;
	if (kvc::parse_key(key, klen, &pk) != 0)
		return SDBV_ERR_BAD_ARG;
	const uint8_t *p = pk.rest, *end = pk.end;
	switch (pk.kind) {
	case 'e': { // He: element id (key) -> SerializedVector (value)
		uint64_t e_id;
		if (!kvc::get_be(p, end, &e_id))
			return SDBV_ERR_BAD_ARG;
		std::vector<float> v;
		if (!kvc::dec_vec_f32(val, val + vlen, v) || v.size() != L->d)
			return SDBV_ERR_UNSUPPORTED;
		L->elems[e_id] = std::move(v);
		return SDBV_OK;
	}
	case 'n': { // Hn: (layer, node) -> edge list
		uint16_t layer;
		uint64_t node;
		if (!kvc::get_be(p, end, &layer) || !kvc::get_be(p, end, &node))
			return SDBV_ERR_BAD_ARG;
		std::vector<uint32_t> edges;
		if (!kvc::dec_node(val, val + vlen, edges))
			return SDBV_ERR_BAD_ARG;
		L->nodes[{layer, node}] = std::move(edges);
		return SDBV_OK;
	}
	case 's': { // Hs: graph state
		if (!kvc::dec_state(val, val + vlen, &L->state))
			return SDBV_ERR_UNSUPPORTED;
		L->has_state = true;
		return SDBV_OK;
	}
	case 'v': { // Hv: escaped vector (key) -> ElementDocs (value)
		std::vector<uint8_t> ser;
		if (!kvc::get_escaped(p, end, ser))
			return SDBV_ERR_BAD_ARG;
		sdbv_kvload::HvEnt ent;
		if (!kvc::dec_vec_f32(ser.data(), ser.data() + ser.size(),
		                      ent.vec) ||
		    ent.vec.size() != L->d)
			return SDBV_ERR_UNSUPPORTED;
		if (!kvc::dec_element_docs(val, val + vlen, &ent.e_id, &ent.docs))
			return SDBV_ERR_UNSUPPORTED;
		L->hv.push_back(std::move(ent));
		return SDBV_OK;
	}
	case 'p': { // Hp: appending id (key) -> VectorPendingUpdate (value)
		uint64_t aid;
		if (!kvc::get_be(p, end, &aid))
			return SDBV_ERR_BAD_ARG;
		kvc::PendingKV pk;
		if (!kvc::dec_pending(val, val + vlen, L->d, &pk))
			return SDBV_ERR_UNSUPPORTED;
		L->hp[aid] = std::move(pk);
		return SDBV_OK;
	}
	default:
		return SDBV_OK; // hd/hi/hh etc.: host-kept, skipped
	}
}

// Builds the host graph from the fed He/Hn/Hs pairs (HnswFlavor::new +
// check_state + HnswLayer::load, layer.rs:504-563 — post-Hl format only).
static int kvload_build_graph(sdbv_kvload *L, sdbv_hnsw **out) {
	if (!L->has_state)
		return SDBV_ERR_BAD_ARG; // an existing index always has Hs
	sdbv_hnsw *h = nullptr;
	int rc = sdbv_hnsw_create(L->ctx, L->d, L->metric, L->m, L->m0, L->efc,
	                          L->extend, L->keep, L->seed, L->ml, &h);
	if (rc)
		return rc;
	uint64_t n = L->state.next_element_id;
	// Untrusted Hs fields bound the allocations below (n*d floats, one
	// n-sized edge vector per layer): reject absurd values up front like
	// the 2^32 doc-id guard, instead of letting bad_alloc abort through
	// the extern "C" boundary.
	if (n > (1ull << 32) || L->state.n_upper_layers > 64) {
		sdbv_hnsw_destroy(h);
		return SDBV_ERR_UNSUPPORTED;
	}
	h->next_id = n;
	h->vecs.assign((size_t)n * L->d, 0.0f);
	h->elem_present.assign(n, 0);
	if (L->metric == SDBV_METRIC_COSINE)
		h->norms.assign(n, 0.0);
	for (auto &e : L->elems) {
		if (e.first >= n) {
			sdbv_hnsw_destroy(h);
			return SDBV_ERR_BAD_ARG;
		}
		std::memcpy(h->vecs.data() + e.first * L->d, e.second.data(),
		            (size_t)L->d * 4);
		h->elem_present[e.first] = 1;
		if (L->metric == SDBV_METRIC_COSINE)
			h->norms[e.first] =
			    sqrt(hnsw::host_sumsq_f32(e.second.data(), L->d));
	}
	uint64_t n_layers = 1 + L->state.n_upper_layers;
	h->layers.clear();
	for (uint64_t l = 0; l < n_layers; l++)
		h->layers.push_back(
		    hnsw::Layer{std::vector<std::vector<uint32_t>>(n),
		                l == 0 ? L->m0 : L->m});
	for (auto &l : h->layers)
		l.in_layer.assign(n, 0);
	for (auto &nd : L->nodes) {
		uint16_t layer = nd.first.first;
		uint64_t node = nd.first.second;
		if (layer >= h->layers.size() || node >= n) {
			sdbv_hnsw_destroy(h);
			return SDBV_ERR_BAD_ARG;
		}
		for (uint32_t e : nd.second)
			if (e >= n) {
				sdbv_hnsw_destroy(h);
				return SDBV_ERR_BAD_ARG;
			}
		// Degree above the declared layer cap is ACCEPTED, like the
		// reference's own loader (layer.rs load reads edge lists with no
		// cap check): it arises legitimately from parameter-mismatched
		// dumps and from this repo's threaded builds (keep-back
		// relaxation). Safe since finalize sizes every per-hop scratch
		// from the ACTUAL max degree (the round-1 advisor's overflow is
		// closed by sizing, its alternative remedy); dec_node already
		// bounds a single node at 65535 edges.
		h->layers[layer].edges[node] = nd.second;
		h->layers[layer].in_layer[node] = 1;
	}
	h->enter_point = L->state.has_ep ? (int64_t)L->state.enter_point : -1;
	h->dirty = true;
	*out = h;
	return SDBV_OK;
}

int sdbv_kvload_finish_hnsw(sdbv_kvload *L, sdbv_hnsw **out) {
	if (!L || !out)
		return SDBV_ERR_BAD_ARG;
	int rc = kvload_build_graph(L, out);
	delete L;
	return rc;
}

int sdbv_kvload_finish_index(sdbv_kvload *L, uint64_t table,
                             sdbv_index **out) {
	if (!L || !out)
		return SDBV_ERR_BAD_ARG;
	sdbv_hnsw *h = nullptr;
	int rc = kvload_build_graph(L, &h);
	if (rc) {
		delete L;
		return rc;
	}
	auto *ix = new sdbv_index();
	ix->h = h;
	ix->table = table;
	uint64_t max_doc = 0;
	bool any_doc = false;
	for (auto &ent : L->hv) {
		std::string key((const char *)ent.vec.data(), (size_t)L->d * 4);
		auto r = ix->vd.emplace(std::move(key),
		                        sdbv_index::ED{(uint32_t)ent.e_id, ent.docs});
		if (r.second)
			ix->by_elem[(uint32_t)ent.e_id] = &r.first->first;
		for (uint64_t d : ent.docs.v) {
			any_doc = true;
			if (d > max_doc)
				max_doc = d;
		}
	}
	// HnswDocsState (hd root) stays host-side, but the allocator is
	// reconstructed EXACTLY up to allocation equivalence. Doc ids
	// allocate densely (docs.rs:78-90), so the reference's persisted
	// state is always {available = [0, next) \ bound, next}; min-first
	// draws make any trailing run of `available` foldable into `next`
	// without changing a single future allocation. Hence: seed
	// available with every id up to the highest Hv-referenced one (Hv
	// references include RECYCLED zombies — the dropped-old-removal
	// quirk — which the reference also holds in `available`), and let
	// sdbv_index_bind_doc_key carve out the live (hi/hd-bound) ids.
	ix->next_doc_id = any_doc ? max_doc + 1 : 0;
	if (ix->next_doc_id > (1ull << 32)) { // corrupt input guard: doc ids
		delete ix;                         // are dense, so this is absurd
		return SDBV_ERR_UNSUPPORTED;
	}
	for (uint64_t dd = 0; dd < ix->next_doc_id; dd++)
		ix->available.insert(dd);
	// outstanding Hp pendings, in appending order (the reference drains
	// the Hp range in key order, index.rs:195-205)
	for (auto &e : L->hp) {
		sdbv_index::Pending p;
		p.kind = e.second.kind;
		p.id = e.second.id;
		p.olds = std::move(e.second.olds);
		p.news = std::move(e.second.news);
		ix->pendings.push_back(std::move(p));
	}
	delete L;
	*out = ix;
	return SDBV_OK;
}

// Export the doc-id <-> record-key-handle map (the host's hi/hd state —
// what it persists so a cold start can re-bind). Returns the entry count;
// writes up to `cap` pairs.
uint64_t sdbv_index_doc_keys(sdbv_index *ix, uint64_t *out_docs,
                             uint64_t *out_keys, uint64_t cap) {
	if (!ix)
		return 0;
	std::lock_guard<std::mutex> lk(ix->mu);
	uint64_t i = 0;
	for (auto &e : ix->doc2key) {
		if (i < cap) {
			out_docs[i] = e.first;
			out_keys[i] = e.second;
		}
		i++;
	}
	return (uint64_t)ix->doc2key.size();
}

// Re-binds one record-key handle to a doc id (the host's hi/hd entries)
// after a cold-start load.
int sdbv_index_bind_doc_key(sdbv_index *ix, uint64_t doc_id,
                            uint64_t record_key) {
	if (!ix)
		return SDBV_ERR_BAD_ARG;
	std::lock_guard<std::mutex> lk(ix->mu);
	ix->key2doc[record_key] = doc_id;
	ix->doc2key[doc_id] = record_key;
	// keep the reconstructed allocator consistent: a bound id is live
	ix->available.erase(doc_id);
	if (doc_id >= ix->next_doc_id) {
		if (doc_id - ix->next_doc_id > (1ull << 32))
			return SDBV_ERR_BAD_ARG; // dense ids: absurd gap = corrupt
		for (uint64_t dd = ix->next_doc_id; dd < doc_id; dd++)
			ix->available.insert(dd);
		ix->next_doc_id = doc_id + 1;
	}
	return SDBV_OK;
}

// ---- KV dump (round-trip + migration tooling) ----

typedef int (*sdbv_kv_write_cb)(void *user, const uint8_t *key, uint64_t klen,
                                const uint8_t *val, uint64_t vlen);

int sdbv_hnsw_dump_kv(sdbv_hnsw *h, uint32_t ns, uint32_t db, const char *tb,
                      uint32_t ix_id, sdbv_kv_write_cb write, void *user) {
	if (!h || !tb || !write)
		return SDBV_ERR_BAD_ARG;
	std::vector<uint8_t> k, v;
	// Hs
	kvc::HnswStateKV st;
	st.has_ep = h->enter_point >= 0;
	st.enter_point = st.has_ep ? (uint64_t)h->enter_point : 0;
	st.next_element_id = h->next_id;
	st.n_upper_layers = h->layers.size() - 1;
	k.clear();
	v.clear();
	kvc::key_hs(k, ns, db, tb, ix_id);
	kvc::enc_state(v, st);
	if (write(user, k.data(), k.size(), v.data(), v.size()))
		return SDBV_ERR_BAD_ARG;
	// He per present element
	for (uint64_t e = 0; e < h->next_id; e++) {
		if (!h->elem_present[e])
			continue;
		k.clear();
		v.clear();
		kvc::key_he(k, ns, db, tb, ix_id, e);
		kvc::enc_vec_f32(v, h->vecs.data() + e * h->d, h->d);
		if (write(user, k.data(), k.size(), v.data(), v.size()))
			return SDBV_ERR_BAD_ARG;
	}
	// Hn per layer member
	for (size_t l = 0; l < h->layers.size(); l++) {
		const hnsw::Layer &layer = h->layers[l];
		for (uint64_t node = 0; node < layer.edges.size(); node++) {
			if (!layer.has((uint32_t)node))
				continue;
			k.clear();
			v.clear();
			kvc::key_hn(k, ns, db, tb, ix_id, (uint16_t)l, node);
			kvc::enc_node(v, layer.edges[node]);
			if (write(user, k.data(), k.size(), v.data(), v.size()))
				return SDBV_ERR_BAD_ARG;
		}
	}
	return SDBV_OK;
}

int sdbv_index_dump_kv(sdbv_index *ix, uint32_t ns, uint32_t db,
                       const char *tb, uint32_t ix_id, sdbv_kv_write_cb write,
                       void *user) {
	if (!ix)
		return SDBV_ERR_BAD_ARG;
	std::lock_guard<std::mutex> lk(ix->mu);
	int rc = sdbv_hnsw_dump_kv(ix->h, ns, db, tb, ix_id, write, user);
	if (rc)
		return rc;
	std::vector<uint8_t> k, v;
	for (auto &e : ix->vd) {
		k.clear();
		v.clear();
		kvc::key_hv(k, ns, db, tb, ix_id, (const float *)e.first.data(),
		            ix->h->d);
		if (!kvc::enc_element_docs(v, e.second.e_id, e.second.docs))
			return SDBV_ERR_UNSUPPORTED; // Ids64 in Bits (roaring) mode
		if (write(user, k.data(), k.size(), v.data(), v.size()))
			return SDBV_ERR_BAD_ARG;
	}
	// outstanding pendings as Hp pairs (RecordKey handles dump as
	// RecordIdKey::Number — see INTEGRATION.md "record-key handles")
	uint64_t aid = 0;
	for (auto &p : ix->pendings) {
		kvc::PendingKV pk;
		pk.kind = p.kind;
		pk.id = p.id;
		pk.olds = p.olds;
		pk.news = p.news;
		k.clear();
		v.clear();
		kvc::key_hp(k, ns, db, tb, ix_id, aid++);
		kvc::enc_pending(v, ix->h->d, pk);
		if (write(user, k.data(), k.size(), v.data(), v.size()))
			return SDBV_ERR_BAD_ARG;
	}
	return SDBV_OK;
}

// check_hnsw_properties (mod.rs:561-570 + the reference index tests'
// element count): present elements == expected == Hv entries, layer edge
// invariants (counts, no self-edges, in-range targets).
int sdbv_index_check_props(sdbv_index *ix, uint64_t expected_count) {
	if (!ix)
		return SDBV_ERR_BAD_ARG;
	std::lock_guard<std::mutex> lk(ix->mu);
	sdbv_hnsw *h = ix->h;
	uint64_t present = 0;
	for (uint8_t p : h->elem_present)
		present += p;
	if (present != expected_count)
		return -10;
	if (present != ix->vd.size())
		return -11;
	for (size_t li = 0; li < h->layers.size(); li++) {
		auto &layer = h->layers[li];
		for (size_t id = 0; id < layer.edges.size(); id++) {
			if (!layer.has((uint32_t)id)) {
				if (!layer.edges[id].empty()) {
					if (getenv("SDBV_DEBUG"))
						fprintf(stderr,
						        "check_props: layer %zu node %zu not "
						        "member, %zu edges (present=%d)\n",
						        li, id, layer.edges[id].size(),
						        id < h->elem_present.size()
						            ? (int)h->elem_present[id]
						            : -1);
					return -12;
				}
				continue;
			}
			if (layer.edges[id].size() > layer.m_max)
				return -13;
			// layer.rs check_props: every graph node is a LIVE element
			// (edge targets may dangle, nodes may not)
			if (id >= h->elem_present.size() || !h->elem_present[id])
				return -15;
			for (uint32_t e : layer.edges[id])
				if (e == id || e >= h->next_id)
					return -14;
		}
	}
	return 0;
}

} // extern "C"
