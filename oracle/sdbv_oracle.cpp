// sdbv_oracle.cpp — TEST INFRASTRUCTURE ONLY.
//
// CPU restatement of the surrealdb/surrealdb vector-KNN hot path, used as the
// parity oracle and as bench.py's cpu_baseline leg (kind "port"). It is NEVER
// the product path: only tests/, __graft_entry__.smoke() and bench.py's
// cpu_baseline may call into this library. The product path is the HIP
// extension (surrealdb_amd/csrc) and must fail loudly when that is missing.
//
// Every function cites the reference file:line it restates (reference:
// /root/reference/surrealdb @ 2026-05-29, Rust — not compilable in this
// environment, no cargo/rustc; see DESIGN.md "Oracle pinning").
//
// Third-party arithmetic restated here (sources NOT vendored in the
// reference; pinned by Cargo.lock):
//  - ndarray 0.17.2 `unrolled_dot` / `unrolled_fold` (numeric_util.rs):
//    eightfold-unrolled accumulation, restated in orc_dot_* / orc_sumsq_*.
//    Call sites: vector.rs:237 (a.dot(b)), vector.rs:238-239 ((a*a).sum()).
//  - ndarray-stats 0.7.0 DeviationExt l2_dist/l1_dist/linf_dist: sequential
//    elementwise folds, final .to_f64().sqrt() for l2. Call sites:
//    vector.rs:282-285 (euclidean), :379-382 (manhattan), :220-229 (chebyshev).
// The reference's own exact-constant tests (vector.rs:723-772) pin results
// across this boundary; residual last-ulp ambiguity (ndarray's pairwise-sum
// threshold for `.sum()`) is documented in DESIGN.md and bounded far below
// the 1e-5 relative score tolerance of the parity contract.
//
// Build: see oracle/Makefile. Compiled WITHOUT -ffast-math: every fp op here
// is IEEE and order-preserving by construction.

#include <cstdint>
#include <cstring>
#include <cfloat>
#include <cmath>
#include <cstdlib>
#include <vector>
#include <deque>
#include <map>
#include <set>
#include <string>
#include <unordered_map>
#include <unordered_set>
#include <algorithm>

#if defined(_OPENMP)
#include <omp.h>
#endif

extern "C" {

// ---------------------------------------------------------------------------
// Deterministic synthetic generator (committed data contract).
// element(seed, gidx) for global element index gidx = row * d + j:
//   splitmix64(seed + gidx) -> u64 -> f64 in [0,1) -> [-20, 20) -> f32.
// Matches the reference's RandomItemGenerator::Float(-20.0, 20.0) by
// distribution (knn.rs:633-643); the exact bit-stream is defined HERE and
// replicated bit-identically by the device generator in
// surrealdb_amd/csrc/scan_kernels.hip.
// ---------------------------------------------------------------------------
static inline uint64_t splitmix64(uint64_t z) {
	z += 0x9E3779B97F4A7C15ULL;
	z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
	z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
	return z ^ (z >> 31);
}

float orc_gen_elem(uint64_t seed, uint64_t gidx) {
	uint64_t x = splitmix64(seed + gidx);
	double u = (double)(x >> 11) * 0x1.0p-53; // [0,1)
	return (float)(-20.0 + 40.0 * u);
}

void orc_gen_f32(uint64_t seed, uint64_t row0, uint64_t nrows, uint32_t d,
                 float *out) {
	for (uint64_t i = 0; i < nrows; i++)
		for (uint32_t j = 0; j < d; j++)
			out[i * d + j] = orc_gen_elem(seed, (row0 + i) * (uint64_t)d + j);
}

// ---------------------------------------------------------------------------
// ndarray 0.17.2 numeric_util::unrolled_dot restatement (f32 / f64).
// Eight independent partials p0..p7, each p += x*y (mul then add, NO fma),
// combined as sum=0; sum+=(p0+p4); sum+=(p1+p5); sum+=(p2+p6); sum+=(p3+p7);
// then sequential tail. Used by Array1::dot on contiguous arrays (no BLAS
// feature in the reference's Cargo.lock).
// ---------------------------------------------------------------------------
#define ORC_UNROLLED_DOT(T)                                                  \
	T p0 = 0, p1 = 0, p2 = 0, p3 = 0, p4 = 0, p5 = 0, p6 = 0, p7 = 0;       \
	uint64_t i = 0;                                                          \
	for (; i + 8 <= d; i += 8) {                                             \
		p0 += a[i + 0] * b[i + 0];                                           \
		p1 += a[i + 1] * b[i + 1];                                           \
		p2 += a[i + 2] * b[i + 2];                                           \
		p3 += a[i + 3] * b[i + 3];                                           \
		p4 += a[i + 4] * b[i + 4];                                           \
		p5 += a[i + 5] * b[i + 5];                                           \
		p6 += a[i + 6] * b[i + 6];                                           \
		p7 += a[i + 7] * b[i + 7];                                           \
	}                                                                        \
	T sum = 0;                                                               \
	sum += (p0 + p4);                                                        \
	sum += (p1 + p5);                                                        \
	sum += (p2 + p6);                                                        \
	sum += (p3 + p7);                                                        \
	for (; i < d; i++)                                                       \
		sum += a[i] * b[i];                                                  \
	return sum;

float orc_dot_f32(const float *a, const float *b, uint64_t d) {
	ORC_UNROLLED_DOT(float)
}
double orc_dot_f64(const double *a, const double *b, uint64_t d) {
	ORC_UNROLLED_DOT(double)
}

// ndarray unrolled_fold restatement for (a*a).sum() (vector.rs:238-239,
// :246-247): same eightfold partials over x*x, combined as
// acc += ((p0+p4)+(p1+p5)); acc += ((p2+p6)+(p3+p7)); sequential tail.
// (ndarray's `.sum()` switches to pairwise blocks above an internal
// threshold; the delta is last-ulp on the norm — see DESIGN.md.)
#define ORC_UNROLLED_SUMSQ(T)                                                \
	T p0 = 0, p1 = 0, p2 = 0, p3 = 0, p4 = 0, p5 = 0, p6 = 0, p7 = 0;       \
	uint64_t i = 0;                                                          \
	for (; i + 8 <= d; i += 8) {                                             \
		p0 += a[i + 0] * a[i + 0];                                           \
		p1 += a[i + 1] * a[i + 1];                                           \
		p2 += a[i + 2] * a[i + 2];                                           \
		p3 += a[i + 3] * a[i + 3];                                           \
		p4 += a[i + 4] * a[i + 4];                                           \
		p5 += a[i + 5] * a[i + 5];                                           \
		p6 += a[i + 6] * a[i + 6];                                           \
		p7 += a[i + 7] * a[i + 7];                                           \
	}                                                                        \
	T acc = 0;                                                               \
	acc += ((p0 + p4) + (p1 + p5));                                          \
	acc += ((p2 + p6) + (p3 + p7));                                          \
	for (; i < d; i++)                                                       \
		acc += a[i] * a[i];                                                  \
	return acc;

float orc_sumsq_f32(const float *a, uint64_t d) { ORC_UNROLLED_SUMSQ(float) }
double orc_sumsq_f64(const double *a, uint64_t d) { ORC_UNROLLED_SUMSQ(double) }

// ---------------------------------------------------------------------------
// Typed distances — Distance::calculate (vector.rs:660-672) on F32/F64.
// metric codes match include/sdbv.h + extensions used only by tests.
// ---------------------------------------------------------------------------
enum {
	M_COSINE = 0,
	M_EUCLIDEAN = 1,
	M_MANHATTAN = 2,
	M_CHEBYSHEV = 3,
	M_HAMMING = 4,
	M_JACCARD = 5,
	M_MINKOWSKI = 6,
	M_PEARSON = 7,
};

// vector.rs:244-249 cosine_distance_f32: f32 dot -> f64; f32 sumsq -> f64 sqrt.
double orc_cosine_f32(const float *a, const float *b, uint64_t d) {
	double dot = (double)orc_dot_f32(a, b, d);
	double na = std::sqrt((double)orc_sumsq_f32(a, d));
	double nb = std::sqrt((double)orc_sumsq_f32(b, d));
	return 1.0 - dot / (na * nb);
}

// vector.rs:236-241 cosine_distance_f64.
double orc_cosine_f64(const double *a, const double *b, uint64_t d) {
	double dot = orc_dot_f64(a, b, d);
	double na = std::sqrt(orc_sumsq_f64(a, d));
	double nb = std::sqrt(orc_sumsq_f64(b, d));
	return 1.0 - dot / (na * nb);
}

// vector.rs:282-283 euclidean via ndarray-stats l2_dist: sequential f32
// accumulation of (a-b)^2, then f64 sqrt (sq_l2_dist returns A, l2_dist
// converts to f64 before sqrt).
double orc_euclidean_f32(const float *a, const float *b, uint64_t d) {
	float acc = 0.0f;
	for (uint64_t i = 0; i < d; i++) {
		float diff = a[i] - b[i];
		acc += diff * diff;
	}
	return std::sqrt((double)acc);
}
double orc_euclidean_f64(const double *a, const double *b, uint64_t d) {
	double acc = 0.0;
	for (uint64_t i = 0; i < d; i++) {
		double diff = a[i] - b[i];
		acc += diff * diff;
	}
	return std::sqrt(acc);
}

// vector.rs:379-382 manhattan via l1_dist (f32 accumulate, then as f64).
static double orc_manhattan_f32(const float *a, const float *b, uint64_t d) {
	float acc = 0.0f;
	for (uint64_t i = 0; i < d; i++)
		acc += std::fabs(a[i] - b[i]);
	return (double)acc;
}
static double orc_manhattan_f64(const double *a, const double *b, uint64_t d) {
	double acc = 0.0;
	for (uint64_t i = 0; i < d; i++)
		acc += std::fabs(a[i] - b[i]);
	return acc;
}

// vector.rs:220-229 chebyshev via linf_dist (max |a-b| in f32, then as f64).
static double orc_chebyshev_f32(const float *a, const float *b, uint64_t d) {
	float m = 0.0f;
	for (uint64_t i = 0; i < d; i++)
		m = std::fmax(m, std::fabs(a[i] - b[i]));
	return (double)m;
}
static double orc_chebyshev_f64(const double *a, const double *b, uint64_t d) {
	double m = 0.0;
	for (uint64_t i = 0; i < d; i++)
		m = std::fmax(m, std::fabs(a[i] - b[i]));
	return m;
}

// vector.rs:292-303 hamming: count of element-wise != (fp compare).
#define ORC_HAMMING_DEF(NAME, T)                                             \
	static double NAME(const T *a, const T *b, uint64_t d) {                 \
		uint64_t acc = 0;                                                    \
		for (uint64_t i = 0; i < d; i++)                                     \
			if (a[i] != b[i])                                                \
				acc++;                                                       \
		return (double)acc;                                                  \
	}
ORC_HAMMING_DEF(orc_hamming_f32, float)
ORC_HAMMING_DEF(orc_hamming_f64, double)

// vector.rs:317-340 jaccard on bit-pattern sets. NOTE the reference asymmetry
// restated faithfully: F64 returns 1 - |I|/|U| (:326), F32 returns |I|/|U|
// (:339).
static double orc_jaccard_f32(const float *a, const float *b, uint64_t d) {
	std::unordered_set<uint32_t> u;
	for (uint64_t i = 0; i < d; i++) {
		uint32_t bits;
		std::memcpy(&bits, &a[i], 4);
		u.insert(bits);
	}
	uint64_t inter = 0;
	for (uint64_t i = 0; i < d; i++) {
		uint32_t bits;
		std::memcpy(&bits, &b[i], 4);
		if (!u.insert(bits).second)
			inter++;
	}
	return (double)inter / (double)u.size();
}
static double orc_jaccard_f64(const double *a, const double *b, uint64_t d) {
	std::unordered_set<uint64_t> u;
	for (uint64_t i = 0; i < d; i++) {
		uint64_t bits;
		std::memcpy(&bits, &a[i], 8);
		u.insert(bits);
	}
	uint64_t inter = 0;
	for (uint64_t i = 0; i < d; i++) {
		uint64_t bits;
		std::memcpy(&bits, &b[i], 8);
		if (!u.insert(bits).second)
			inter++;
	}
	return 1.0 - (double)inter / (double)u.size();
}

// vector.rs:389-399 minkowski: per-element to_float() -> f64 diff, abs, powf,
// sequential sum, final pow(1/order).
#define ORC_MINKOWSKI_DEF(NAME, T)                                           \
	static double NAME(const T *a, const T *b, uint64_t d, double order) {   \
		double acc = 0.0;                                                    \
		for (uint64_t i = 0; i < d; i++)                                     \
			acc += std::pow(std::fabs((double)a[i] - (double)b[i]), order);  \
		return std::pow(acc, 1.0 / order);                                   \
	}
ORC_MINKOWSKI_DEF(orc_minkowski_f32, float)
ORC_MINKOWSKI_DEF(orc_minkowski_f64, double)

// vector.rs:413-440 pearson (similarity used directly as the metric value).
// mean: ndarray mean = sum/len in the element type, then .to_float().
static double orc_pearson_f32(const float *a, const float *b, uint64_t d) {
	float sa = 0.0f, sb = 0.0f;
	{ // ndarray .mean() -> .sum()/len in f32; restated as unrolled_fold sum
		const float *x = a;
		float p0 = 0, p1 = 0, p2 = 0, p3 = 0, p4 = 0, p5 = 0, p6 = 0, p7 = 0;
		uint64_t i = 0;
		for (; i + 8 <= d; i += 8) {
			p0 += x[i]; p1 += x[i + 1]; p2 += x[i + 2]; p3 += x[i + 3];
			p4 += x[i + 4]; p5 += x[i + 5]; p6 += x[i + 6]; p7 += x[i + 7];
		}
		sa += ((p0 + p4) + (p1 + p5));
		sa += ((p2 + p6) + (p3 + p7));
		for (; i < d; i++) sa += x[i];
	}
	{
		const float *x = b;
		float p0 = 0, p1 = 0, p2 = 0, p3 = 0, p4 = 0, p5 = 0, p6 = 0, p7 = 0;
		uint64_t i = 0;
		for (; i + 8 <= d; i += 8) {
			p0 += x[i]; p1 += x[i + 1]; p2 += x[i + 2]; p3 += x[i + 3];
			p4 += x[i + 4]; p5 += x[i + 5]; p6 += x[i + 6]; p7 += x[i + 7];
		}
		sb += ((p0 + p4) + (p1 + p5));
		sb += ((p2 + p6) + (p3 + p7));
		for (; i < d; i++) sb += x[i];
	}
	double mx = (double)(sa / (float)d);
	double my = (double)(sb / (float)d);
	double sum_xy = 0, sum_x2 = 0, sum_y2 = 0;
	for (uint64_t i = 0; i < d; i++) {
		double dx = (double)a[i] - mx, dy = (double)b[i] - my;
		sum_xy += dx * dy;
		sum_x2 += dx * dx;
		sum_y2 += dy * dy;
	}
	double den = std::sqrt(sum_x2 * sum_y2);
	if (den == 0.0) return 0.0;
	return sum_xy / den;
}
static double orc_pearson_f64(const double *a, const double *b, uint64_t d) {
	double sa = 0, sb = 0;
	for (uint64_t i = 0; i < d; i++) sa += a[i];
	for (uint64_t i = 0; i < d; i++) sb += b[i];
	double mx = sa / (double)d, my = sb / (double)d;
	double sum_xy = 0, sum_x2 = 0, sum_y2 = 0;
	for (uint64_t i = 0; i < d; i++) {
		double dx = a[i] - mx, dy = b[i] - my;
		sum_xy += dx * dy;
		sum_x2 += dx * dx;
		sum_y2 += dy * dy;
	}
	double den = std::sqrt(sum_x2 * sum_y2);
	if (den == 0.0) return 0.0;
	return sum_xy / den;
}

// Distance::calculate dispatch (vector.rs:660-672) for F32 / F64 vectors.
double orc_dist_f32(uint8_t metric, double order, const float *a,
                    const float *b, uint64_t d) {
	switch (metric) {
	case M_COSINE: return orc_cosine_f32(a, b, d);
	case M_EUCLIDEAN: return orc_euclidean_f32(a, b, d);
	case M_MANHATTAN: return orc_manhattan_f32(a, b, d);
	case M_CHEBYSHEV: return orc_chebyshev_f32(a, b, d);
	case M_HAMMING: return orc_hamming_f32(a, b, d);
	case M_JACCARD: return orc_jaccard_f32(a, b, d);
	case M_MINKOWSKI: return orc_minkowski_f32(a, b, d, order);
	case M_PEARSON: return orc_pearson_f32(a, b, d);
	}
	return NAN;
}
double orc_dist_f64(uint8_t metric, double order, const double *a,
                    const double *b, uint64_t d) {
	switch (metric) {
	case M_COSINE: return orc_cosine_f64(a, b, d);
	case M_EUCLIDEAN: return orc_euclidean_f64(a, b, d);
	case M_MANHATTAN: return orc_manhattan_f64(a, b, d);
	case M_CHEBYSHEV: return orc_chebyshev_f64(a, b, d);
	case M_HAMMING: return orc_hamming_f64(a, b, d);
	case M_JACCARD: return orc_jaccard_f64(a, b, d);
	case M_MINKOWSKI: return orc_minkowski_f64(a, b, d, order);
	case M_PEARSON: return orc_pearson_f64(a, b, d);
	}
	return NAN;
}

// ---------------------------------------------------------------------------
// Number-path distances — Distance::compute (catalog/schema/index.rs:287-301
// -> fnc/util/math/vector.rs) over Vec<Number> where every element is
// Number::Float(f64): plain sequential f64 arithmetic.
//  - cosine: dot (fnc vector.rs:279-281, sequential sum of products, the
//    Number Sum starts at Int(0) and promotes exactly) / magnitudes
//    (:301-313, sequential sum of squares, f64 sqrt).
//  - euclidean: (:288-298) sequential sum of (a-b)^2, f64 sqrt.
// ---------------------------------------------------------------------------
double orc_dist_number(uint8_t metric, double order, const double *a,
                       const double *b, uint64_t d) {
	switch (metric) {
	case M_COSINE: {
		double dot = 0, ma = 0, mb = 0;
		for (uint64_t i = 0; i < d; i++) dot += a[i] * b[i];
		for (uint64_t i = 0; i < d; i++) ma += a[i] * a[i];
		for (uint64_t i = 0; i < d; i++) mb += b[i] * b[i];
		return 1.0 - dot / (std::sqrt(ma) * std::sqrt(mb));
	}
	case M_EUCLIDEAN: {
		double acc = 0;
		for (uint64_t i = 0; i < d; i++) {
			double diff = a[i] - b[i];
			acc += diff * diff;
		}
		return std::sqrt(acc);
	}
	case M_MANHATTAN: { // fnc vector.rs:153-158
		double acc = 0;
		for (uint64_t i = 0; i < d; i++) acc += std::fabs(a[i] - b[i]);
		return acc;
	}
	case M_CHEBYSHEV: { // fnc vector.rs:218-228 (fold starts at f64::MIN)
		double m = -DBL_MAX;
		for (uint64_t i = 0; i < d; i++) m = std::fmax(m, std::fabs(a[i] - b[i]));
		return m;
	}
	case M_MINKOWSKI: {
		double acc = 0;
		for (uint64_t i = 0; i < d; i++)
			acc += std::pow(std::fabs(a[i] - b[i]), order);
		return std::pow(acc, 1.0 / order);
	}
	case M_HAMMING: return orc_hamming_f64(a, b, d);
	case M_JACCARD: { // fnc vector.rs:120-126: Number-equality sets, sim only
		std::unordered_set<uint64_t> u;
		for (uint64_t i = 0; i < d; i++) {
			uint64_t bits; std::memcpy(&bits, &a[i], 8); u.insert(bits);
		}
		uint64_t inter = 0;
		for (uint64_t i = 0; i < d; i++) {
			uint64_t bits; std::memcpy(&bits, &b[i], 8);
			if (!u.insert(bits).second) inter++;
		}
		return (double)inter / (double)u.size();
	}
	case M_PEARSON: { // fnc vector.rs:132-147
		double sa = 0, sb = 0;
		for (uint64_t i = 0; i < d; i++) sa += a[i];
		for (uint64_t i = 0; i < d; i++) sb += b[i];
		double m1 = sa / (double)d, m2 = sb / (double)d;
		double covar = 0;
		for (uint64_t i = 0; i < d; i++) covar += (a[i] - m1) * (b[i] - m2);
		covar /= (double)d;
		double v1 = 0, v2 = 0;
		for (uint64_t i = 0; i < d; i++) v1 += (a[i] - m1) * (a[i] - m1);
		for (uint64_t i = 0; i < d; i++) v2 += (b[i] - m2) * (b[i] - m2);
		double s1 = std::sqrt(v1 / (double)d), s2 = std::sqrt(v2 / (double)d);
		return covar / (s1 * s2);
	}
	}
	return NAN;
}

// ---------------------------------------------------------------------------
// f64::total_cmp sort key (knn.rs:128-160 FloatKey): IEEE-754 totalOrder.
// Monotone u64 key: negative -> ~bits, else bits | sign.
// ---------------------------------------------------------------------------
uint64_t orc_total_key(double x) {
	uint64_t bits;
	std::memcpy(&bits, &x, 8);
	if (bits >> 63)
		return ~bits;
	return bits | 0x8000000000000000ULL;
}

// ---------------------------------------------------------------------------
// Brute-force top-K — KnnTopK semantics (exec/operators/knn_topk.rs:166-267):
// bounded heap of size k; when full, insert only if STRICTLY closer than the
// current worst (ties keep the earlier row, :216-227); final order ascending
// (distance, insertion seq) (:61-73, :230-236). For ordinal ids this equals
// the KnnResultBuilder ordering (knn.rs:363-437): (dist total_cmp asc, id asc).
// mode: 0 = typed-F32 distances (Distance::calculate), 1 = Number-path on the
// same values widened to f64 (Distance::compute).
// ---------------------------------------------------------------------------
struct TopK {
	uint32_t k;
	// sorted ascending by (total_key(dist), seq)
	std::vector<std::pair<std::pair<uint64_t, uint64_t>, double>> v;
	void push(double dist, uint64_t seq) {
		std::pair<uint64_t, uint64_t> key{orc_total_key(dist), seq};
		if (v.size() >= k) {
			// knn_topk.rs:218-219: insert only if dist < worst dist
			// (strict, distance only — seq never displaces on tie)
			if (!(dist < v.back().second))
				return;
			v.pop_back();
		}
		auto it = std::lower_bound(
		    v.begin(), v.end(), key,
		    [](const auto &e, const std::pair<uint64_t, uint64_t> &kk) {
			    return e.first < kk;
		    });
		v.insert(it, {key, dist});
	}
};

void orc_topk_f32(uint8_t metric, double order, const float *corpus,
                  uint64_t n, uint32_t d, const float *q, uint32_t k,
                  uint64_t *out_ids, double *out_dists, uint32_t *out_n) {
	TopK t{k, {}};
	for (uint64_t r = 0; r < n; r++)
		t.push(orc_dist_f32(metric, order, q, corpus + r * d, d), r);
	*out_n = (uint32_t)t.v.size();
	for (size_t i = 0; i < t.v.size(); i++) {
		out_ids[i] = t.v[i].first.second;
		out_dists[i] = t.v[i].second;
	}
}

void orc_topk_number(uint8_t metric, double order, const double *corpus,
                     uint64_t n, uint32_t d, const double *q, uint32_t k,
                     uint64_t *out_ids, double *out_dists, uint32_t *out_n) {
	TopK t{k, {}};
	for (uint64_t r = 0; r < n; r++)
		t.push(orc_dist_number(metric, order, q, corpus + r * d, d), r);
	*out_n = (uint32_t)t.v.size();
	for (size_t i = 0; i < t.v.size(); i++) {
		out_ids[i] = t.v[i].first.second;
		out_dists[i] = t.v[i].second;
	}
}

// Multithreaded (OpenMP) variant for bench.py's cpu_baseline leg: row-chunked,
// per-thread TopK, merged with the same (dist, id) ascending contract.
// Returns threads used.
int orc_topk_f32_mt(uint8_t metric, double order, const float *corpus,
                    uint64_t n, uint32_t d, const float *q, uint32_t k,
                    uint64_t *out_ids, double *out_dists, uint32_t *out_n,
                    int nthreads) {
#if defined(_OPENMP)
	if (nthreads <= 0)
		nthreads = omp_get_max_threads();
	std::vector<TopK> parts((size_t)nthreads, TopK{k, {}});
#pragma omp parallel num_threads(nthreads)
	{
		int tid = omp_get_thread_num();
		TopK &t = parts[(size_t)tid];
#pragma omp for schedule(static)
		for (int64_t r = 0; r < (int64_t)n; r++)
			t.push(orc_dist_f32(metric, order, q, corpus + (uint64_t)r * d, d),
			       (uint64_t)r);
	}
	std::vector<std::pair<double, uint64_t>> all;
	for (auto &p : parts)
		for (auto &e : p.v)
			all.push_back({e.second, e.first.second});
	std::sort(all.begin(), all.end(), [](const auto &x, const auto &y) {
		auto kx = std::make_pair(orc_total_key(x.first), x.second);
		auto ky = std::make_pair(orc_total_key(y.first), y.second);
		return kx < ky;
	});
	uint32_t m = (uint32_t)std::min<size_t>(k, all.size());
	*out_n = m;
	for (uint32_t i = 0; i < m; i++) {
		out_ids[i] = all[i].second;
		out_dists[i] = all[i].first;
	}
	return nthreads;
#else
	orc_topk_f32(metric, order, corpus, n, d, q, k, out_ids, out_dists, out_n);
	return 1;
#endif
}

// ---------------------------------------------------------------------------
// HNSW restatement — hnsw/mod.rs + layer.rs + heuristic.rs.
// Element vectors are F32 (the reference's default VectorType, define.rs:1107).
// DynamicSet is restated as an insertion-ordered vector set (the reference's
// ArraySet semantics, dynamicset.rs; its AHashSet variant iterates in
// nondeterministic order, so graph-level bitwise parity is not defined by the
// reference itself — parity is judged on result sets/recall, matching the
// reference's own tests hnsw/mod.rs:1104-1184).
// ---------------------------------------------------------------------------

struct OrcPQ {
	// DoublePriorityQueue (knn.rs:15-123): BTreeMap<total_cmp(dist), FIFO
	// deque>. Restated as a sorted (total_key, push-seq) vector with a
	// lazy head — the exact observable order (FIFO within equal distance
	// on pop_first; latest-of-max-key on pop_last), pinned verbatim by the
	// reference's own test_double_priority_queue in
	// tests/test_pq_semantics.py.
	struct E {
		uint64_t key;
		uint32_t seq;
		uint64_t id;
		double d;
	};
	std::vector<E> v; // ascending (key, seq); v[head..] is the live queue
	size_t head = 0;
	uint32_t next_seq = 0;
	size_t n = 0;
	void push(double d, uint64_t id) {
		E e{orc_total_key(d), next_seq++, id, d};
		auto it = std::upper_bound(
		    v.begin() + head, v.end(), e, [](const E &a, const E &b) {
			    return a.key != b.key ? a.key < b.key : a.seq < b.seq;
		    });
		v.insert(it, e);
		n++;
	}
	bool pop_first(double *d, uint64_t *id) {
		if (n == 0) return false;
		*d = v[head].d;
		*id = v[head].id;
		head++;
		n--;
		return true;
	}
	bool pop_last(double *d, uint64_t *id) {
		if (n == 0) return false;
		*d = v.back().d;
		*id = v.back().id;
		v.pop_back();
		n--;
		return true;
	}
	bool peek_first(double *d, uint64_t *id) const {
		if (n == 0) return false;
		*d = v[head].d;
		*id = v[head].id;
		return true;
	}
	double peek_last_dist(double fallback) const {
		return n == 0 ? fallback : v.back().d;
	}
	std::vector<std::pair<double, uint64_t>> to_vec() const {
		std::vector<std::pair<double, uint64_t>> out;
		out.reserve(n);
		for (size_t i = head; i < v.size(); i++)
			out.push_back({v[i].d, v[i].id});
		return out;
	}
};

struct OrcLayer {
	// UndirectedGraph over insertion-ordered sets (graph.rs, dynamicset.rs).
	// `in_layer` mirrors the reference's explicit nodes map (graph.rs:
	// get_edges returns None for absent nodes; remove_node_and_bidirectional_
	// edges needs membership, not just an empty edge list).
	std::vector<std::vector<uint32_t>> edges; // indexed by element id
	uint32_t m_max;
	std::vector<uint8_t> in_layer;
	bool has(uint64_t id) const {
		return id < in_layer.size() && in_layer[id];
	}
	void add_node(uint64_t id) {
		if (edges.size() <= id)
			edges.resize(id + 1);
		if (in_layer.size() <= id)
			in_layer.resize(id + 1, 0);
		in_layer[id] = 1;
	}
};

struct orc_hnsw {
	uint32_t d;
	uint8_t metric;
	double order;
	uint32_t m, m0, efc;
	bool extend, keep;
	double ml;
	uint64_t rng_state;
	std::vector<float> vecs;          // element id -> row (id * d)
	std::vector<uint8_t> present;     // per-layer membership: layers[l] set
	std::vector<OrcLayer> layers;     // [0] = layer0 (m0), [1..] upper (m)
	int64_t enter_point = -1;
	uint64_t next_id = 0;
	// per-element max layer (for membership checks)
	std::vector<int32_t> top_layer;
	// elements map membership (hnsw/elements.rs: remove() deletes the entry;
	// vector storage is retained, the slot just becomes dead)
	std::vector<uint8_t> elem_present;
};

// are_all_docs_in_pending (layer.rs:320-338) needs the index's vec_docs;
// searches below thread this through. `pending` = the DocId bitmap returned
// by search_pendings; `docs_of_elem` resolves an element to its doc set.
struct OrcPend {
	const std::set<uint64_t> *pending;
	const void *idx; // orc_index*
	bool (*all_docs_pending)(const void *idx, uint64_t e_id,
	                         const std::set<uint64_t> *pending);
};

static double hdist(const orc_hnsw *h, const float *a, const float *b) {
	return orc_dist_f32(h->metric, h->order, a, b, h->d);
}
static const float *hvec(const orc_hnsw *h, uint64_t id) {
	return h->vecs.data() + id * h->d;
}

orc_hnsw *orc_hnsw_new(uint32_t d, uint8_t metric, double order, uint32_t m,
                       uint32_t m0, uint32_t efc, int extend, int keep,
                       uint64_t rng_seed, double ml) {
	auto *h = new orc_hnsw();
	h->d = d; h->metric = metric; h->order = order;
	h->m = m; h->m0 = m0; h->efc = efc;
	h->extend = extend != 0; h->keep = keep != 0;
	h->ml = ml;
	h->rng_state = rng_seed;
	h->layers.push_back(OrcLayer{{}, m0});
	return h;
}
void orc_hnsw_free(orc_hnsw *h) { delete h; }

// hnsw/mod.rs:263-266 get_random_level: floor(-ln(U) * ml), U uniform [0,1).
// RNG restated with splitmix64 (the reference seeds SmallRng from thread_rng —
// nondeterministic; level SEQUENCE distribution is what matters).
static uint32_t next_level(orc_hnsw *h) {
	h->rng_state = splitmix64(h->rng_state);
	double u = (double)(h->rng_state >> 11) * 0x1.0p-53;
	if (u <= 0.0) u = 0x1.0p-53;
	double lvl = std::floor(-std::log(u) * h->ml);
	if (lvl < 0) lvl = 0;
	if (lvl > 30) lvl = 30;
	return (uint32_t)lvl;
}

// layer.rs:184-223 HnswLayer::search — the best-first ef-bounded loop.
// `pend` restates the pending_docs parameter: an element whose docs are ALL
// pending is excluded from `candidates` (the expansion frontier) but still
// pushed into `w` (layer.rs:209-212 — the reference pushes to w outside the
// exclusion check; restated as-is).
static void layer_search(const orc_hnsw *h, const OrcLayer &layer,
                         const float *q, OrcPQ &candidates,
                         std::unordered_set<uint64_t> &visited, OrcPQ &w,
                         uint32_t ef, const OrcPend *pend = nullptr) {
	double fq_dist = w.peek_last_dist(1.7976931348623157e308);
	double cq_dist; uint64_t doc;
	while (candidates.pop_first(&cq_dist, &doc)) {
		if (cq_dist > fq_dist)
			break;
		if (layer.has(doc)) {
			for (uint32_t e_id : layer.edges[doc]) {
				if (!visited.insert(e_id).second)
					continue;
				// elements.get_vector -> None for removed elements
				// (layer.rs:206: dangling edges — left behind by
				// insert-time pruning asymmetry — are skipped AFTER the
				// visited mark)
				if (e_id < h->elem_present.size() &&
				    !h->elem_present[e_id])
					continue;
				double e_dist = hdist(h, hvec(h, e_id), q);
				if (e_dist < fq_dist || w.n < ef) {
					if (!pend || !pend->all_docs_pending(pend->idx, e_id,
					                                     pend->pending))
						candidates.push(e_dist, e_id);
					w.push(e_dist, e_id);
					if (w.n > ef) {
						double dd; uint64_t ii;
						w.pop_last(&dd, &ii);
					}
					fq_dist = w.peek_last_dist(1.7976931348623157e308);
				}
			}
		}
	}
}

// heuristic.rs:193-216 is_closer.
static bool is_closer(const orc_hnsw *h, double e_dist, uint64_t e_id,
                      std::vector<uint32_t> &r) {
	const float *ev = hvec(h, e_id);
	for (uint32_t r_id : r) {
		double r_dist = hdist(h, ev, hvec(h, r_id));
		if (e_dist > r_dist)
			return false;
	}
	r.push_back((uint32_t)e_id);
	return true;
}

// heuristic.rs:61-82 heuristic / :84-116 heuristic_keep.
static void heuristic_select(const orc_hnsw *h, const OrcLayer &layer,
                             OrcPQ c, std::vector<uint32_t> &res, bool keep) {
	uint32_t m_max = layer.m_max;
	if (c.n <= m_max) {
		for (auto &e : c.to_vec())
			res.push_back((uint32_t)e.second);
		return;
	}
	std::vector<uint64_t> pruned;
	double e_dist; uint64_t e_id;
	while (c.pop_first(&e_dist, &e_id)) {
		if (is_closer(h, e_dist, e_id, res)) {
			if (res.size() == m_max)
				break;
		} else if (keep) {
			pruned.push_back(e_id);
		}
	}
	if (keep) {
		size_t nmore = m_max - res.size();
		for (size_t i = 0; i < nmore && i < pruned.size(); i++)
			res.push_back((uint32_t)pruned[i]);
	}
}

// heuristic.rs:118-157 extend_candidates (`ignore`: heuristic.rs:130-134 —
// the element being removed is excluded from the extension set).
static void extend_candidates(const orc_hnsw *h, const OrcLayer &layer,
                              uint64_t q_id, const float *q_pt, OrcPQ &c,
                              int64_t ignore = -1) {
	std::unordered_set<uint64_t> ex;
	for (auto &e : c.to_vec())
		ex.insert(e.second);
	if (ignore >= 0)
		ex.insert((uint64_t)ignore);
	std::vector<std::pair<double, uint64_t>> ext;
	for (auto &e : c.to_vec()) {
		uint64_t e_id = e.second;
		if (e_id < layer.edges.size()) {
			for (uint32_t e_adj : layer.edges[e_id]) {
				if (e_adj != q_id && ex.insert(e_adj).second) {
					// get_distance -> None for removed elements
					// (heuristic.rs:140-143: ex is marked first)
					if (e_adj < h->elem_present.size() &&
					    !h->elem_present[e_adj])
						continue;
					double dd = hdist(h, q_pt, hvec(h, e_adj));
					ext.push_back({dd, e_adj});
				}
			}
		}
	}
	for (auto &e : ext)
		c.push(e.first, e.second);
}

static void select_neighbors(const orc_hnsw *h, const OrcLayer &layer,
                             uint64_t q_id, const float *q_pt, OrcPQ c,
                             std::vector<uint32_t> &res, int64_t ignore = -1) {
	if (h->extend)
		extend_candidates(h, layer, q_id, q_pt, c, ignore);
	heuristic_select(h, layer, std::move(c), res, h->keep);
}

// layer.rs:342-387 HnswLayer::insert (+ graph.rs add_node_and_bidirectional_edges)
static OrcPQ layer_insert(orc_hnsw *h, OrcLayer &layer, uint32_t q_id,
                          const float *q_pt, OrcPQ eps) {
	// search_multi (layer.rs:151-161)
	OrcPQ w = eps;
	std::unordered_set<uint64_t> visited;
	for (auto &e : eps.to_vec())
		visited.insert(e.second);
	layer_search(h, layer, q_pt, eps, visited, w, h->efc);
	OrcPQ out_eps = w;

	std::vector<uint32_t> neighbors;
	select_neighbors(h, layer, q_id, q_pt, w, neighbors);

	// add node + bidirectional edges (graph.rs:52-64 — the back-edge
	// `nodes.entry(e).or_insert_with(..)` IMPLICITLY creates a missing
	// target node, e.g. an upper-layer seed that was never inserted here)
	layer.add_node(q_id);
	layer.edges[q_id] = neighbors;
	for (uint32_t e_id : neighbors) {
		layer.add_node(e_id);
		layer.edges[e_id].push_back(q_id);
	}

	// prune over-full neighbors (layer.rs:363-377)
	for (uint32_t e_id : neighbors) {
		auto &conn = layer.edges[e_id];
		if (conn.size() > layer.m_max) {
			const float *e_pt = hvec(h, e_id);
			OrcPQ e_c; // build_priority_list (layer.rs:390-404):
			           // get_vector -> None skips removed elements, so a
			           // prune also cleanses dangling edges
			for (uint32_t n_id : conn) {
				if (n_id < h->elem_present.size() &&
				    !h->elem_present[n_id])
					continue;
				e_c.push(hdist(h, e_pt, hvec(h, n_id)), n_id);
			}
			std::vector<uint32_t> e_new;
			select_neighbors(h, layer, e_id, e_pt, std::move(e_c), e_new);
			layer.edges[e_id] = e_new;
		}
	}
	return out_eps;
}

void orc_hnsw_insert_level(orc_hnsw *h, const float *pt, uint32_t q_level) {
	uint32_t q_id = (uint32_t)h->next_id++;
	h->vecs.insert(h->vecs.end(), pt, pt + h->d);
	h->top_layer.push_back((int32_t)q_level);
	h->elem_present.push_back(1);
	uint32_t top_up_layers = (uint32_t)h->layers.size() - 1;

	for (uint32_t i = top_up_layers; i < q_level; i++)
		h->layers.push_back(OrcLayer{{}, h->m});

	if (h->enter_point < 0) {
		// insert_first_element (hnsw/mod.rs:272-291)
		for (uint32_t l = 0; l <= q_level && l < (uint32_t)h->layers.size(); l++) {
			h->layers[l].add_node(q_id);
			h->layers[l].edges[q_id] = {};
		}
		h->enter_point = (int64_t)q_id;
		return;
	}

	// insert_element (hnsw/mod.rs:296-378)
	uint64_t ep_id = (uint64_t)h->enter_point;
	double ep_dist = hdist(h, pt, hvec(h, ep_id));
	if (q_level < top_up_layers) {
		for (uint32_t l = top_up_layers; l > q_level; l--) {
			// upper layers are h->layers[1..]; layer index l == graph layer l
			OrcPQ cand; cand.push(ep_dist, ep_id);
			std::unordered_set<uint64_t> visited{ep_id};
			OrcPQ w = cand;
			layer_search(h, h->layers[l], pt, cand, visited, w, 1);
			double dd; uint64_t ii;
			if (w.peek_first(&dd, &ii)) { ep_dist = dd; ep_id = ii; }
		}
	}
	OrcPQ eps; eps.push(ep_dist, ep_id);
	uint32_t insert_to = std::min(q_level, top_up_layers);
	for (uint32_t l = insert_to; l >= 1; l--)
		eps = layer_insert(h, h->layers[l], q_id, pt, std::move(eps));
	layer_insert(h, h->layers[0], q_id, pt, std::move(eps));

	for (uint32_t l = top_up_layers + 1; l <= q_level; l++) {
		h->layers[l].add_node(q_id);
		h->layers[l].edges[q_id] = {};
	}
	if (q_level > top_up_layers)
		h->enter_point = (int64_t)q_id;
}

void orc_hnsw_insert(orc_hnsw *h, const float *pt) {
	orc_hnsw_insert_level(h, pt, next_level(h));
}

// layer.rs:92-108 search_single_with_ignore — seeds the search FROM the
// ignored element itself (its distance enters candidates but visited blocks
// it from w); returns the closest found element or -1 (None).
static int64_t search_single_with_ignore(const orc_hnsw *h,
                                         const OrcLayer &layer,
                                         const float *pt, uint64_t ignore_id,
                                         uint32_t ef) {
	std::unordered_set<uint64_t> visited{ignore_id};
	OrcPQ candidates;
	candidates.push(hdist(h, pt, hvec(h, ignore_id)), ignore_id);
	OrcPQ w;
	layer_search(h, layer, pt, candidates, visited, w, ef);
	double dd; uint64_t ii;
	if (w.peek_first(&dd, &ii))
		return (int64_t)ii;
	return -1;
}

// layer.rs:164-181 search_multi_with_ignore.
static OrcPQ search_multi_with_ignore(const orc_hnsw *h, const OrcLayer &layer,
                                      const float *pt,
                                      const std::vector<uint64_t> &ignore_ids,
                                      uint32_t efc) {
	OrcPQ candidates;
	for (uint64_t id : ignore_ids)
		candidates.push(hdist(h, pt, hvec(h, id)), id);
	std::unordered_set<uint64_t> visited(ignore_ids.begin(),
	                                     ignore_ids.end());
	OrcPQ w;
	layer_search(h, layer, pt, candidates, visited, w, efc);
	return w;
}

// layer.rs:408-460 HnswLayer::remove: drop the node and its back-edges
// (graph.rs remove_node_and_bidirectional_edges), then repair each former
// neighbour with an efc-search (ignoring itself and the removed element) and
// a heuristic re-selection (ignore = the removed element).
static bool layer_remove(orc_hnsw *h, OrcLayer &layer, uint64_t e_id) {
	if (!layer.has(e_id))
		return false;
	std::vector<uint32_t> f_ids = std::move(layer.edges[e_id]);
	layer.edges[e_id].clear();
	layer.in_layer[e_id] = 0;
	for (uint32_t f : f_ids) {
		auto &fe = layer.edges[f];
		fe.erase(std::remove(fe.begin(), fe.end(), (uint32_t)e_id),
		         fe.end());
	}
	for (uint32_t q_id : f_ids) {
		const float *q_pt = hvec(h, q_id);
		OrcPQ c = search_multi_with_ignore(h, layer, q_pt, {q_id, e_id},
		                                   h->efc);
		std::vector<uint32_t> q_new_conn;
		select_neighbors(h, layer, q_id, q_pt, std::move(c), q_new_conn,
		                 (int64_t)e_id);
		layer.edges[q_id] = q_new_conn; // graph.set_node
	}
	return true;
}

// hnsw/mod.rs:398-455 Hnsw::remove. Returns 1 if the element was removed.
int orc_hnsw_remove(orc_hnsw *h, uint64_t e_id) {
	if (e_id >= h->next_id || !h->elem_present[e_id])
		return 0; // elements.get_vector -> None (mod.rs:401)
	bool removed = false;
	const float *e_pt = hvec(h, e_id);
	int64_t new_enter_point =
	    ((int64_t)e_id == h->enter_point) ? -1 : h->enter_point;
	// upper layers, top-down (mod.rs:411-422)
	for (size_t l = h->layers.size() - 1; l >= 1; l--) {
		if (new_enter_point < 0)
			new_enter_point = search_single_with_ignore(
			    h, h->layers[l], e_pt, e_id, h->efc);
		if (layer_remove(h, h->layers[l], e_id))
			removed = true;
	}
	// possible new enter_point at layer0 (mod.rs:424-429)
	if (new_enter_point < 0)
		new_enter_point =
		    search_single_with_ignore(h, h->layers[0], e_pt, e_id, h->efc);
	if (layer_remove(h, h->layers[0], e_id))
		removed = true;
	h->elem_present[e_id] = 0; // elements.remove (mod.rs:448)
	h->enter_point = new_enter_point;
	return removed ? 1 : 0;
}

// hnsw/mod.rs:459-482 knn_search + :521-548 search_ep, with the optional
// pending_docs parameter threaded through every layer (index.rs knn path).
static uint32_t hnsw_search_core(orc_hnsw *h, const float *q, uint32_t k,
                                 uint32_t ef, const OrcPend *pend,
                                 uint64_t *out_ids, double *out_dists) {
	if (h->enter_point < 0)
		return 0;
	uint64_t ep_id = (uint64_t)h->enter_point;
	double ep_dist = hdist(h, q, hvec(h, ep_id));
	for (size_t l = h->layers.size() - 1; l >= 1; l--) {
		OrcPQ cand; cand.push(ep_dist, ep_id);
		std::unordered_set<uint64_t> visited{ep_id};
		OrcPQ w = cand;
		layer_search(h, h->layers[l], q, cand, visited, w, 1, pend);
		double dd; uint64_t ii;
		if (w.peek_first(&dd, &ii)) { ep_dist = dd; ep_id = ii; }
	}
	OrcPQ cand; cand.push(ep_dist, ep_id);
	std::unordered_set<uint64_t> visited{ep_id};
	OrcPQ w = cand;
	layer_search(h, h->layers[0], q, cand, visited, w, ef, pend);
	// to_vec_limit(k) (knn.rs:92-104)
	auto v = w.to_vec();
	uint32_t nout = (uint32_t)std::min<size_t>(k, v.size());
	for (uint32_t i = 0; i < nout; i++) {
		out_dists[i] = v[i].first;
		out_ids[i] = v[i].second;
	}
	return nout;
}

uint32_t orc_hnsw_search(orc_hnsw *h, const float *q, uint32_t k, uint32_t ef,
                         uint64_t *out_ids, double *out_dists) {
	return hnsw_search_core(h, q, k, ef, nullptr, out_ids, out_dists);
}

uint32_t orc_hnsw_num_layers(orc_hnsw *h) { return (uint32_t)h->layers.size(); }
uint64_t orc_hnsw_num_elements(orc_hnsw *h) { return h->next_id; }
int64_t orc_hnsw_entry_point(orc_hnsw *h) { return h->enter_point; }

// check_hnsw_props (hnsw/mod.rs:561-570): edge count <= m_max, no self-edges.
// Returns 0 if OK, else a negative code.
int orc_hnsw_check_props(orc_hnsw *h) {
	for (size_t l = 0; l < h->layers.size(); l++) {
		const OrcLayer &layer = h->layers[l];
		for (size_t id = 0; id < layer.edges.size(); id++) {
			if (!layer.has(id)) {
				if (!layer.edges[id].empty())
					return -3; // absent nodes have no edge list
				continue;
			}
			if (layer.edges[id].size() > layer.m_max + 0)
				return -1;
			for (uint32_t e : layer.edges[id])
				if (e == id)
					return -2;
			// layer.rs check_props: every graph node is a LIVE element
			if (id >= h->elem_present.size() || !h->elem_present[id])
				return -4;
		}
	}
	return 0;
}

// Export layer-0 CSR for the GPU HNSW path (offsets: n+1, edges concatenated).
uint64_t orc_hnsw_l0_edge_count(orc_hnsw *h) {
	uint64_t c = 0;
	for (auto &e : h->layers[0].edges)
		c += e.size();
	return c;
}
void orc_hnsw_l0_export(orc_hnsw *h, uint32_t *offsets, uint32_t *edges) {
	uint32_t off = 0;
	for (uint64_t i = 0; i < h->next_id; i++) {
		offsets[i] = off;
		if (i < h->layers[0].edges.size())
			for (uint32_t e : h->layers[0].edges[i])
				edges[off++] = e;
	}
	offsets[h->next_id] = off;
}

// Import a prebuilt graph (vectors + per-layer adjacency) so the oracle
// can SEARCH a graph built elsewhere — the bench's cpu_baseline leg runs
// orc_hnsw_search on the very graph the GPU searches (test infrastructure,
// like everything in this file). Layers load via orc_hnsw_import_layer.
orc_hnsw *orc_hnsw_import(uint32_t d, uint8_t metric, double order,
                          uint32_t m, uint32_t m0, uint32_t efc,
                          uint64_t n, const float *vecs, int64_t enter_point,
                          uint32_t nlayers) {
	auto *h = orc_hnsw_new(d, metric, order, m, m0, efc, 0, 0, 0, 0.0);
	h->vecs.assign(vecs, vecs + n * d);
	h->next_id = n;
	h->elem_present.assign(n, 1);
	h->top_layer.assign(n, 0);
	h->enter_point = enter_point;
	h->layers.clear();
	for (uint32_t l = 0; l < (nlayers ? nlayers : 1); l++) {
		OrcLayer L{std::vector<std::vector<uint32_t>>(n), l == 0 ? m0 : m};
		L.in_layer.assign(n, 0);
		h->layers.push_back(std::move(L));
	}
	return h;
}
int orc_hnsw_import_layer(orc_hnsw *h, uint32_t l, const uint32_t *offsets,
                          const uint32_t *edges, const uint8_t *in_layer) {
	if (!h || l >= h->layers.size())
		return -1;
	OrcLayer &L = h->layers[l];
	for (uint64_t i = 0; i < h->next_id; i++) {
		L.in_layer[i] = in_layer[i];
		L.edges[i].assign(edges + offsets[i], edges + offsets[i + 1]);
		if (in_layer[i] && (int32_t)l > h->top_layer[i])
			h->top_layer[i] = (int32_t)l;
	}
	return 0;
}

// Host-side upper-layer descent (search_ep, hnsw/mod.rs:521-548) for the GPU
// path: returns the layer-0 entry element and its distance.
void orc_hnsw_search_ep(orc_hnsw *h, const float *q, uint64_t *ep_id_out,
                        double *ep_dist_out) {
	uint64_t ep_id = (uint64_t)h->enter_point;
	double ep_dist = hdist(h, q, hvec(h, ep_id));
	for (size_t l = h->layers.size() - 1; l >= 1; l--) {
		OrcPQ cand; cand.push(ep_dist, ep_id);
		std::unordered_set<uint64_t> visited{ep_id};
		OrcPQ w = cand;
		layer_search(h, h->layers[l], q, cand, visited, w, 1);
		double dd; uint64_t ii;
		if (w.peek_first(&dd, &ii)) { ep_dist = dd; ep_id = ii; }
	}
	*ep_id_out = ep_id;
	*ep_dist_out = ep_dist;
}

const float *orc_hnsw_vec_ptr(orc_hnsw *h, uint64_t id) { return hvec(h, id); }

// ===========================================================================
// Index layer — HnswIndex (hnsw/index.rs) + VecDocs/Ids64 (docs.rs, knn.rs)
// + the pendings queue (VectorPendingUpdate over Hp keys), restated as an
// in-memory structure: everything the reference keeps behind the KV
// transaction (hi/hd record-key maps, Hp pendings stream, Hv vector->docs
// entries) is held in maps here; the record-key itself is an opaque u64
// handle supplied by the host (INTEGRATION.md "record-key handles").
// ===========================================================================

// knn.rs:163-326 Ids64 — doc-id set with size-dependent representation.
// Vec1..Vec8 keep INSERTION order; the 9th insert collapses to Bits
// (RoaringTreemap — iteration ascending); dropping back to exactly 8 keeps
// ascending order. insert()/remove() return "a new variant was produced"
// exactly like the reference, because VecDocs only persists on Some — Bits
// in-place mutations are DROPPED by the caller (docs.rs:374-382, :437-447;
// restated as-is, quirks included — see test_index_oracle.py).
struct OrcIds64 {
	std::vector<uint64_t> v;
	bool bits = false;
	size_t len() const { return v.size(); }
	bool empty() const { return v.empty(); }
	bool contains(uint64_t d) const {
		return std::find(v.begin(), v.end(), d) != v.end();
	}
	// knn.rs:234-256: None on duplicate; Bits insert mutates in place and
	// returns None (=> false here).
	bool insert_ret_variant(uint64_t d) {
		if (contains(d))
			return false;
		if (!bits) {
			v.push_back(d);
			if (v.size() > 8) { // Vec8 -> Bits: RoaringTreemap::from(sorted)
				std::sort(v.begin(), v.end());
				bits = true;
			}
			return true; // Vec*->Vec* and Vec8->Bits both produce Some
		}
		v.insert(std::lower_bound(v.begin(), v.end(), d), d);
		return false; // Bits: in-place, None
	}
	// knn.rs:258-326: per-variant removal. Returns true and writes *out if a
	// NEW variant was produced; Bits removals mutate in place (and only
	// produce a variant when dropping to exactly 8). The Vec2 non-member
	// case reproduces the reference's `find(|i| i != d).map(One)` exactly.
	bool remove_ret_variant(uint64_t d, OrcIds64 *out) {
		if (bits) {
			auto it = std::lower_bound(v.begin(), v.end(), d);
			bool had = (it != v.end() && *it == d);
			if (had)
				v.erase(it); // RoaringTreemap::remove, in place
			if (!had || v.size() != 8)
				return false;
			out->v = v; // Bits -> Vec8 (ascending, b.iter() order)
			out->bits = false;
			return true;
		}
		switch (v.size()) {
		case 0:
			return false; // Empty -> None
		case 1:
			if (v[0] == d) { // One -> Empty
				out->v.clear();
				out->bits = false;
				return true;
			}
			return false;
		case 2:
			// Vec2: first element != d becomes One (knn.rs:268 — for a
			// non-member d this DROPS the second element; restated as-is)
			for (uint64_t x : v)
				if (x != d) {
					out->v = {x};
					out->bits = false;
					return true;
				}
			return false;
		default: {
			// Vec3..Vec8: filter; a variant is produced only when exactly
			// one element was removed
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

// knn.rs:363-437 KnnResultBuilder: BTreeSet<(FloatKey, VectorId)> +
// per-VectorId multiplicity. VectorId (hnsw/mod.rs VectorId enum, derived
// Ord): kind 0 = DocId(u64) < kind 1 = RecordKey (host-ordered u64 handle).
struct OrcVid {
	uint8_t kind;
	uint64_t id;
	bool operator<(const OrcVid &o) const {
		if (kind != o.kind)
			return kind < o.kind;
		return id < o.id;
	}
};
struct OrcBuilder {
	size_t knn;
	struct Ent {
		uint64_t key; // orc_total_key(dist)
		OrcVid vid;
		double dist;
		bool operator<(const Ent &o) const {
			if (key != o.key)
				return key < o.key;
			return vid < o.vid;
		}
	};
	std::set<Ent> pl;
	std::map<OrcVid, size_t> count;
	explicit OrcBuilder(size_t k) : knn(k) {}
	// knn.rs:386-394 check_add: plain f64 `>` against the current worst
	// (NOT total_cmp — NaN submitted passes).
	bool check_add(double submitted) const {
		if (pl.size() >= knn && !pl.empty() &&
		    submitted > std::prev(pl.end())->dist)
			return false;
		return true;
	}
	// knn.rs:410-433 add_vector_id_result. Returns true + *evicted when an
	// id fell out of the result entirely (its count hit 0).
	bool add(double dist, OrcVid vid, OrcVid *evicted) {
		pl.insert(Ent{orc_total_key(dist), vid, dist});
		count[vid]++; // incremented even when the set insert was a dup
		if (pl.size() <= knn)
			return false;
		auto last = std::prev(pl.end());
		OrcVid ev = last->vid;
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
	// knn.rs:398-407 add_graph_result: one add per doc in the Ids64.
	void add_graph(double dist, const OrcIds64 &docs) {
		for (uint64_t doc : docs.v)
			add(dist, OrcVid{0, doc}, nullptr);
	}
};

// docs.rs:281-450 VecDocs (non-hashed Hv path — the default; the hashed Hh
// variant differs only in KV key layout, not in observable results) plus
// docs.rs:20-135 HnswDocs (record-key <-> doc-id maps, recycled allocation).
struct OrcED {
	uint64_t e_id;
	OrcIds64 docs;
};
struct orc_index {
	orc_hnsw *h;
	// Hv entries: serialized vector bytes -> (element, docs)
	std::unordered_map<std::string, OrcED> vd;
	std::unordered_map<uint64_t, const std::string *> by_elem;
	// HnswDocs (hi/hd keys + HnswDocsState)
	std::map<uint64_t, uint64_t> key2doc, doc2key;
	std::set<uint64_t> available;
	uint64_t next_doc_id = 0;
	// Hp pendings, appending order (index.rs:131-174)
	struct Pending {
		uint8_t kind; // 0 DocId, 1 RecordKey
		uint64_t id;
		std::vector<float> olds, news; // n*d each
	};
	std::vector<Pending> pendings;
};

static std::string vec_key(const orc_index *ix, const float *v) {
	return std::string((const char *)v, (size_t)ix->h->d * 4);
}

orc_index *orc_index_new(uint32_t d, uint8_t metric, double order, uint32_t m,
                         uint32_t m0, uint32_t efc, int extend, int keep,
                         uint64_t seed, double ml) {
	auto *ix = new orc_index();
	ix->h = orc_hnsw_new(d, metric, order, m, m0, efc, extend, keep, seed, ml);
	return ix;
}
void orc_index_free(orc_index *ix) {
	if (!ix)
		return;
	orc_hnsw_free(ix->h);
	delete ix;
}
orc_hnsw *orc_index_hnsw(orc_index *ix) { return ix->h; }
uint64_t orc_index_doc_count(orc_index *ix) { return ix->doc2key.size(); }
uint64_t orc_index_pending_count(orc_index *ix) {
	return ix->pendings.size();
}

// HnswIndex::index (index.rs:138-186): resolve the id kind via the hi map
// (HnswDocs::get_doc_id) and append one VectorPendingUpdate. old/new vectors
// are n*d f32 each (the host's content_to_vectors output).
int orc_index_enqueue(orc_index *ix, uint64_t record_key, const float *olds,
                      uint32_t n_old, const float *news, uint32_t n_new) {
	orc_index::Pending p;
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
	return 0;
}

// docs.rs:64-76 resolve + :78-90 next_doc_id (smallest recycled id first).
static uint64_t docs_resolve(orc_index *ix, uint64_t record_key) {
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
static void docs_remove(orc_index *ix, uint64_t doc_id) {
	auto it = ix->doc2key.find(doc_id);
	if (it == ix->doc2key.end())
		return;
	ix->key2doc.erase(it->second);
	ix->doc2key.erase(it);
	ix->available.insert(doc_id);
}

// docs.rs:363-393 VecDocs::insert.
static void vd_insert(orc_index *ix, const float *vec, uint64_t doc_id) {
	std::string key = vec_key(ix, vec);
	auto it = ix->vd.find(key);
	if (it == ix->vd.end()) {
		// new vector: insert into the graph, ElementDocs::new
		uint32_t e_id = (uint32_t)ix->h->next_id;
		orc_hnsw_insert(ix->h, vec);
		auto r = ix->vd.emplace(std::move(key), OrcED{e_id, {}});
		r.first->second.docs.v = {doc_id};
		ix->by_elem[e_id] = &r.first->first;
	} else {
		// existing vector: persist only if a new Ids64 variant was
		// produced (docs.rs:376-381 — Bits in-place adds are dropped)
		OrcED ed = it->second; // owned copy, like tx.get
		if (ed.docs.insert_ret_variant(doc_id))
			it->second = ed;
	}
}

// docs.rs:424-449 VecDocs::remove.
static void vd_remove(orc_index *ix, const float *vec, uint64_t doc_id) {
	std::string key = vec_key(ix, vec);
	auto it = ix->vd.find(key);
	if (it == ix->vd.end())
		return;
	OrcED ed = it->second; // owned copy, like tx.get
	OrcIds64 new_docs;
	if (ed.docs.remove_ret_variant(doc_id, &new_docs)) {
		if (new_docs.empty()) {
			uint64_t e_id = ed.e_id;
			ix->by_elem.erase(e_id);
			ix->vd.erase(it);
			orc_hnsw_remove(ix->h, e_id);
		} else {
			ed.docs = new_docs;
			it->second = ed;
		}
	}
	// else: no variant produced — mutation dropped (docs.rs:437-447)
}

// HnswIndex::index_pendings + index_pending (index.rs:188-257): drain the
// queue in appending order; old-vector removals only for DocId pendings;
// empty new_vectors deletes the doc mapping; RecordKey ids resolve (and
// allocate) at apply time.
uint64_t orc_index_apply(orc_index *ix) {
	uint64_t count = 0;
	uint32_t d = ix->h->d;
	for (auto &p : ix->pendings) {
		if (p.kind == 0) {
			for (size_t i = 0; i * d < p.olds.size(); i++)
				vd_remove(ix, p.olds.data() + i * d, p.id);
			if (p.news.empty())
				docs_remove(ix, p.id);
		}
		if (!p.news.empty()) {
			uint64_t doc_id = (p.kind == 0) ? p.id : docs_resolve(ix, p.id);
			for (size_t i = 0; i * d < p.news.size(); i++)
				vd_insert(ix, p.news.data() + i * d, doc_id);
		}
		count++;
	}
	ix->pendings.clear();
	return count;
}

// layer.rs:320-338 are_all_docs_in_pending: an element whose vector has no
// VecDocs entry, or whose every doc is pending, counts as all-pending.
static bool orc_all_docs_pending(const void *vix, uint64_t e_id,
                                 const std::set<uint64_t> *pending) {
	if (!pending || pending->empty())
		return false;
	auto *ix = (const orc_index *)vix;
	auto it = ix->by_elem.find(e_id);
	if (it != ix->by_elem.end()) {
		const OrcIds64 &docs = ix->vd.at(*it->second).docs;
		for (uint64_t doc : docs.v)
			if (!pending->count(doc))
				return false;
	}
	return true;
}

// HnswIndex::knn_search (index.rs:270-335): search_pendings overlay, then
// the graph search excluding all-pending elements (the layer.rs:209-212
// candidates-only exclusion), doc expansion via VecDocs, one KnnResultBuilder
// across both. Out arrays sized k; returns the entry count. No cond_filter
// here (hnsw/filter.rs is SURVEY $8f rank 2).
uint32_t orc_index_knn(orc_index *ix, const float *q, uint32_t k, uint32_t ef,
                       uint8_t *out_kinds, uint64_t *out_ids,
                       double *out_dists) {
	orc_hnsw *h = ix->h;
	uint32_t d = h->d;
	OrcBuilder builder(k);
	// search_pendings (index.rs:366-421): two passes over the queue
	std::set<uint64_t> all_existing;
	std::map<OrcVid, const std::vector<float> *> non_deleted;
	for (auto &p : ix->pendings) {
		if (p.kind == 0)
			all_existing.insert(p.id);
		OrcVid vid{p.kind, p.id};
		if (p.news.empty())
			non_deleted.erase(vid);
		else
			non_deleted[vid] = &p.news;
	}
	if (!(all_existing.empty() && non_deleted.empty())) {
		for (auto &e : non_deleted) {
			const std::vector<float> &vecs = *e.second;
			for (size_t i = 0; i * d < vecs.size(); i++) {
				// Distance::calculate on the typed F32 path
				// (idx/trees/vector.rs:660-672)
				double dd = orc_dist_f32(h->metric, h->order, q,
				                         vecs.data() + i * d, d);
				if (builder.check_add(dd))
					builder.add(dd, e.first, nullptr);
			}
		}
	}
	// graph search with the pending_docs bitmap (None when no DocId pending)
	OrcPend pend{&all_existing, ix, &orc_all_docs_pending};
	const OrcPend *pp = all_existing.empty() ? nullptr : &pend;
	std::vector<uint64_t> gids(std::max<uint32_t>(k, ef));
	std::vector<double> gdists(std::max<uint32_t>(k, ef));
	uint32_t ng = hnsw_search_core(h, q, k, ef, pp, gids.data(),
	                               gdists.data());
	// add_graph_results (index.rs:454-483)
	for (uint32_t i = 0; i < ng; i++) {
		uint64_t e_id = gids[i];
		if (!builder.check_add(gdists[i]))
			continue;
		auto it = ix->by_elem.find(e_id);
		if (it == ix->by_elem.end())
			continue; // get_vector/get_docs -> None
		builder.add_graph(gdists[i], ix->vd.at(*it->second).docs);
	}
	// collect() -> ascending (FloatKey, VectorId)
	uint32_t n = 0;
	for (const auto &e : builder.pl) {
		out_kinds[n] = e.vid.kind;
		out_ids[n] = e.vid.id;
		out_dists[n] = e.dist;
		n++;
	}
	return n;
}

// ---------------------------------------------------------------------------
// Filtered KNN (hnsw/filter.rs + layer.rs:110-318 + the index.rs filtered
// flow). The WHERE-condition evaluation itself (is_record_truthy:
// KV record fetch + expression compute, filter.rs:111-138) stays on the
// host side of the boundary as a callback; the library keeps the
// reference's FilterCache semantics (one evaluation per VectorId while
// cached, entries expired when the builder evicts the id) and all the
// accept/expand gating. The callback must be deterministic within one call.
// ---------------------------------------------------------------------------

typedef int (*orc_truthy_cb)(void *user, uint8_t kind, uint64_t id);
typedef void (*orc_expire_cb)(void *user, uint8_t kind, uint64_t id);

struct OrcFilter {
	orc_truthy_cb cb;
	orc_expire_cb ex;
	void *user;
	// FilterCache (filter.rs:22): VectorId -> truthy
	std::map<std::pair<uint8_t, uint64_t>, bool> cache;
	bool truthy(uint8_t kind, uint64_t id) {
		auto key = std::make_pair(kind, id);
		auto it = cache.find(key);
		if (it != cache.end())
			return it->second; // filter.rs:77-78 cached
		bool t = cb(user, kind, id) != 0;
		cache[key] = t;
		return t;
	}
	void expire(uint8_t kind, uint64_t id) { // filter.rs:141-144
		cache.erase({kind, id});
		if (ex)
			ex(user, kind, id);
	}
	// filter.rs:53-66 check_any_doc_truthy (Ids64 iteration order)
	bool any_doc_truthy(const OrcIds64 &docs) {
		for (uint64_t d : docs.v)
			if (truthy(0, d))
				return true;
		return false;
	}
};

// layer.rs:308-318 check_all_docs_in_pending (the plain Ids64 variant used
// by add_if_truthy — distinct from are_all_docs_in_pending).
static bool check_all_docs_in_pending(const OrcIds64 &docs,
                                      const std::set<uint64_t> *pending) {
	if (!pending || pending->empty())
		return false;
	for (uint64_t d : docs.v)
		if (!pending->count(d))
			return false;
	return true;
}

// layer.rs:278-306 add_if_truthy: w gets the element only if its vector has
// a VecDocs entry, the docs are not all-pending, and >=1 doc is truthy.
// Looks docs up BY VECTOR (vec_docs.get_docs(e_pt)).
static bool add_if_truthy(orc_index *ix, uint32_t ef, OrcPQ &w,
                          const float *e_pt, double e_dist, uint64_t e_id,
                          OrcFilter &filter,
                          const std::set<uint64_t> *pending) {
	auto it = ix->vd.find(vec_key(ix, e_pt));
	if (it == ix->vd.end())
		return false;
	const OrcIds64 &docs = it->second.docs;
	if (check_all_docs_in_pending(docs, pending))
		return false;
	if (filter.any_doc_truthy(docs)) {
		w.push(e_dist, e_id);
		if (w.n > ef) {
			double dd;
			uint64_t ii;
			w.pop_last(&dd, &ii);
		}
		return true;
	}
	return false;
}

// layer.rs:226-275 search_with_filter: candidates expand unconditionally
// (within the distance gate); w is gated by add_if_truthy.
static void layer_search_with_filter(orc_index *ix, const OrcLayer &layer,
                                     const float *q, OrcPQ &candidates,
                                     std::unordered_set<uint64_t> &visited,
                                     OrcPQ &w, uint32_t ef,
                                     OrcFilter &filter,
                                     const std::set<uint64_t> *pending) {
	orc_hnsw *h = ix->h;
	double f_dist = w.peek_last_dist(1.7976931348623157e308);
	double cq_dist;
	uint64_t doc;
	while (candidates.pop_first(&cq_dist, &doc)) {
		if (cq_dist > f_dist)
			break;
		if (!layer.has(doc))
			continue;
		for (uint32_t e_id : layer.edges[doc]) {
			if (!visited.insert(e_id).second)
				continue;
			if (e_id < h->elem_present.size() && !h->elem_present[e_id])
				continue; // get_vector -> None
			double e_dist = hdist(h, hvec(h, e_id), q);
			if (e_dist < f_dist || w.n < ef) {
				candidates.push(e_dist, e_id);
				if (add_if_truthy(ix, ef, w, hvec(h, e_id), e_dist, e_id,
				                  filter, pending))
					f_dist = w.peek_last_dist(1.7976931348623157e308);
			}
		}
	}
}

// HnswIndex::knn_search with cond_filter (index.rs:270-335 +
// knn_search_with_filter mod.rs:484-515 + search_single_with_filter
// layer.rs:110-149). Returns entry count (<= k).
uint32_t orc_index_knn_filtered(orc_index *ix, const float *q, uint32_t k,
                                uint32_t ef, orc_truthy_cb truthy,
                                orc_expire_cb expire, void *user,
                                uint8_t *out_kinds, uint64_t *out_ids,
                                double *out_dists) {
	orc_hnsw *h = ix->h;
	uint32_t d = h->d;
	OrcBuilder builder(k);
	OrcFilter filter{truthy, expire, user, {}};
	// search_pendings with the filter (index.rs:400-404): non-truthy
	// pending ids are skipped before their vectors are scored
	std::set<uint64_t> all_existing;
	std::map<OrcVid, const std::vector<float> *> non_deleted;
	for (auto &p : ix->pendings) {
		if (p.kind == 0)
			all_existing.insert(p.id);
		OrcVid vid{p.kind, p.id};
		if (p.news.empty())
			non_deleted.erase(vid);
		else
			non_deleted[vid] = &p.news;
	}
	if (!(all_existing.empty() && non_deleted.empty())) {
		for (auto &e : non_deleted) {
			if (!filter.truthy(e.first.kind, e.first.id))
				continue;
			const std::vector<float> &vecs = *e.second;
			for (size_t i = 0; i * d < vecs.size(); i++) {
				double dd = orc_dist_f32(h->metric, h->order, q,
				                         vecs.data() + i * d, d);
				if (builder.check_add(dd)) {
					OrcVid ev;
					if (builder.add(dd, e.first, &ev))
						filter.expire(ev.kind, ev.id);
				}
			}
		}
	}
	const std::set<uint64_t> *pp =
	    all_existing.empty() ? nullptr : &all_existing;
	OrcPend pend{&all_existing, ix, &orc_all_docs_pending};
	const OrcPend *ep_pend = all_existing.empty() ? nullptr : &pend;
	// knn_search_with_filter (mod.rs:484-515)
	std::vector<std::pair<double, uint64_t>> neighbors;
	if (h->enter_point >= 0) {
		// search_ep: plain upper-layer descent with pending (mod.rs:520-548)
		uint64_t ep_id = (uint64_t)h->enter_point;
		double ep_dist = hdist(h, q, hvec(h, ep_id));
		for (size_t l = h->layers.size() - 1; l >= 1; l--) {
			OrcPQ cand;
			cand.push(ep_dist, ep_id);
			std::unordered_set<uint64_t> visited{ep_id};
			OrcPQ w = cand;
			layer_search(h, h->layers[l], q, cand, visited, w, 1, ep_pend);
			double dd;
			uint64_t ii;
			if (w.peek_first(&dd, &ii)) {
				ep_dist = dd;
				ep_id = ii;
			}
		}
		// search_single_with_filter (layer.rs:110-149). NOTE the seed
		// passes SEARCH.PT as e_pt (layer.rs:125-135): the entry point
		// enters w only if the QUERY VECTOR itself has a VecDocs entry
		// with a truthy, not-all-pending doc — restated as-is.
		OrcPQ candidates;
		candidates.push(ep_dist, ep_id);
		std::unordered_set<uint64_t> visited{ep_id};
		OrcPQ w;
		add_if_truthy(ix, ef, w, q, ep_dist, ep_id, filter, pp);
		layer_search_with_filter(ix, h->layers[0], q, candidates, visited,
		                         w, ef, filter, pp);
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
			OrcVid ev;
			if (builder.add(nb.first, OrcVid{0, docid}, &ev))
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
	return n;
}

// Test hooks: a standalone OrcLayer driven by the reference's
// UndirectedGraph operations (graph.rs:43-130), so tests can replay
// graph.rs:166-233 test_undirected_graph verbatim. The internal build /
// remove paths use the same primitives (add_node, bidirectional add,
// set_node on members, remove+back-edge cleanup).
void *orc_test_graph_new(uint32_t m_max) {
	auto *l = new OrcLayer();
	l->m_max = m_max;
	return l;
}
void orc_test_graph_free(void *g) { delete (OrcLayer *)g; }
// graph.rs:43-50 add_empty_node: 1 if inserted, 0 if it already existed
int orc_test_graph_add_empty_node(void *g, uint64_t id) {
	auto *l = (OrcLayer *)g;
	if (l->has(id))
		return 0;
	l->add_node(id);
	l->edges[id].clear();
	return 1;
}
// graph.rs:52-64 add_node_and_bidirectional_edges (implicit target create)
void orc_test_graph_add_bidir(void *g, uint64_t id, const uint64_t *edges,
                              uint32_t n) {
	auto *l = (OrcLayer *)g;
	for (uint32_t i = 0; i < n; i++) {
		l->add_node(edges[i]);
		l->edges[edges[i]].push_back((uint32_t)id);
	}
	l->add_node(id);
	l->edges[id].clear();
	for (uint32_t i = 0; i < n; i++)
		l->edges[id].push_back((uint32_t)edges[i]);
}
// graph.rs:66-68 set_node (creates a missing node)
void orc_test_graph_set_node(void *g, uint64_t id, const uint64_t *edges,
                             uint32_t n) {
	auto *l = (OrcLayer *)g;
	l->add_node(id);
	l->edges[id].clear();
	for (uint32_t i = 0; i < n; i++)
		l->edges[id].push_back((uint32_t)edges[i]);
}
// graph.rs:70-81 remove_node_and_bidirectional_edges: returns the removed
// node's edge count (into out) or -1 (None)
int orc_test_graph_remove(void *g, uint64_t id, uint64_t *out,
                          uint32_t cap) {
	auto *l = (OrcLayer *)g;
	if (!l->has(id))
		return -1;
	std::vector<uint32_t> f = l->edges[id];
	l->edges[id].clear();
	l->in_layer[id] = 0;
	for (uint32_t e : f) {
		auto &fe = l->edges[e];
		fe.erase(std::remove(fe.begin(), fe.end(), (uint32_t)id), fe.end());
	}
	uint32_t m = std::min<uint32_t>((uint32_t)f.size(), cap);
	for (uint32_t i = 0; i < m; i++)
		out[i] = f[i];
	return (int)f.size();
}
// edge readout: -1 if the node is absent (get_edges None), else count
int orc_test_graph_edges(void *g, uint64_t id, uint64_t *out, uint32_t cap) {
	auto *l = (OrcLayer *)g;
	if (!l->has(id))
		return -1;
	const auto &e = l->edges[id];
	uint32_t m = std::min<uint32_t>((uint32_t)e.size(), cap);
	for (uint32_t i = 0; i < m; i++)
		out[i] = e[i];
	return (int)e.size();
}

// Test hooks: drive OrcPQ directly so tests can replay the reference's own
// test_double_priority_queue sequence (knn.rs:735-790).
void *orc_test_pq_new() { return new OrcPQ(); }
void orc_test_pq_free(void *q) { delete (OrcPQ *)q; }
uint64_t orc_test_pq_len(void *q) { return ((OrcPQ *)q)->n; }
void orc_test_pq_push(void *q, double d, uint64_t id) {
	((OrcPQ *)q)->push(d, id);
}
int orc_test_pq_peek_first(void *q, double *d, uint64_t *id) {
	return ((OrcPQ *)q)->peek_first(d, id) ? 1 : 0;
}
int orc_test_pq_peek_last_dist(void *q, double *d) {
	auto *pq = (OrcPQ *)q;
	if (pq->n == 0)
		return 0;
	*d = pq->peek_last_dist(0);
	return 1;
}
int orc_test_pq_pop_first(void *q, double *d, uint64_t *id) {
	return ((OrcPQ *)q)->pop_first(d, id) ? 1 : 0;
}
int orc_test_pq_pop_last(void *q, double *d, uint64_t *id) {
	auto *pq = (OrcPQ *)q;
	if (pq->n == 0)
		return 0;
	return pq->pop_last(d, id) ? 1 : 0;
}

// Test hook: drive OrcIds64 directly so tests can restate the reference's
// own test_ids sequence (knn.rs:669-717) — variant transitions (Some/None)
// and contents/order, bit for bit.
OrcIds64 *orc_ids64_new() { return new OrcIds64(); }
void orc_ids64_free(OrcIds64 *s) { delete s; }
// returns 1 if a new variant was produced (the reference's Some)
int orc_ids64_insert(OrcIds64 *s, uint64_t d) {
	return s->insert_ret_variant(d) ? 1 : 0;
}
// returns 1 if a new variant was produced; the set is REPLACED by it (the
// caller-persists model) — matching how VecDocs uses the return value
int orc_ids64_remove(OrcIds64 *s, uint64_t d) {
	OrcIds64 out;
	if (s->remove_ret_variant(d, &out)) {
		*s = out;
		return 1;
	}
	return 0;
}
uint32_t orc_ids64_export(OrcIds64 *s, uint64_t *out, int *is_bits) {
	for (size_t i = 0; i < s->v.size(); i++)
		out[i] = s->v[i];
	*is_bits = s->bits ? 1 : 0;
	return (uint32_t)s->v.size();
}

// check_hnsw_properties for the index (mod.rs:561-570 + the element count
// from the reference's index tests): elements present == expected, plus the
// layer invariants.
int orc_index_check_props(orc_index *ix, uint64_t expected_count) {
	uint64_t present = 0;
	for (uint8_t p : ix->h->elem_present)
		present += p;
	if (present != expected_count)
		return -10;
	if (present != ix->vd.size())
		return -11; // every element is exactly one Hv entry
	return orc_hnsw_check_props(ix->h);
}

} // extern "C"
