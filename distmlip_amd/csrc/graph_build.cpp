// graph_build.cpp — native neighbor list + slab partitioner (CPU, OpenMP).
//
// From-scratch implementation of the graph-layer CONTRACT of the reference
// (AegisIK/DistMLIP distributed/fpis.c + subgraph_creation_utils.c); see
// include/distmlip_graph.h for the per-function reference citations.  The
// implementation is new: count+fill prefix-sum passes over centers (thread-
// count-independent deterministic output), a single expanded-image cell
// grid, sequential per-partition passes with reusable global->local maps.
//
// Semantics intentionally preserved from the reference, including quirks:
//  * walls at frac_min + i*(len/P) + 1e-10, nudged by 1e-10 on exact
//    collision with an atom coordinate (utils.c:1431-1455)
//  * which_partition: first wall strictly greater than the coordinate
//    (utils.c:1312-1322)
//  * width check uses walls[0] (NOT the slab width) times the norm of
//    lattice COLUMN dim (utils.c:1512-1529) and hard-fails (-4)
//  * each border atom may be owed to exactly ONE foreign partition; a
//    violation prints a warning and keeps the last assignment
//    (utils.c:1243-1251)
//  * line edges skip backtracks by ATOM id only (utils.c:727-729)
//  * no self edges in any periodic image (fpis.c:833)
// Divergences (documented in DESIGN.md): num_partitions == 1 allowed;
// per-BDE global edge ids exported (ghost geometry computed locally).

#include "../../include/distmlip_graph.h"

#include <algorithm>
#include <array>
#include <atomic>
#include <cmath>
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#include <omp.h>

#include <malloc.h>
#include <cstdlib>

#define DM_T(name) \
    do { if (getenv("DM_TIMING")) { \
        std::fprintf(stderr, "DM_TIMING %-22s %.1f ms\n", name, \
                     (omp_get_wtime() - t_phase_) * 1e3); \
        t_phase_ = omp_get_wtime(); } } while (0)

namespace {

// The builder allocates multi-GB arrays per call; with glibc's default
// mmap threshold every call mmaps fresh pages and every free munmaps them,
// so each rebuild pays full page-fault cost (~1 s/GB).  Raise the
// thresholds once so freed arena memory is REUSED across rebuilds.
struct MallocTuning {
    MallocTuning() {
        mallopt(M_MMAP_THRESHOLD, 1 << 30);
        mallopt(M_TRIM_THRESHOLD, 1 << 30);
    }
} g_malloc_tuning;

thread_local std::string g_err;

void set_err(const std::string& s) { g_err = s; }

constexpr double WALL_EPSILON = 1e-10;  // utils.c:9

// avoid value-initialization of multi-GB output arrays (they are fully
// written by the fill passes; zeroing them first doubles memory traffic)
template <class T>
struct NoInit : std::allocator<T> {
    template <class U> struct rebind { using other = NoInit<U>; };
    template <class U, class... A>
    void construct(U* p, A&&... a) {
        if constexpr (sizeof...(A) > 0) ::new ((void*)p) U(std::forward<A>(a)...);
    }
};
template <class T> using rvec = std::vector<T, NoInit<T>>;

struct Part {
    std::vector<int64_t> markers, line_markers, map_de, map_ude;
    rvec<int64_t> global_ids, edge_gids, bde_edge_gids;
    // hot device-bound arrays are int32 (n < 2^31) so the Python layer
    // uploads them without a host-side conversion pass
    rvec<int32_t> src_local, dst_local, line_src, line_dst, line_center;
    rvec<int32_t> row_ptr, src_perm, src_row_ptr;
    rvec<int32_t> line_row_ptr, line_src_perm, line_src_row_ptr;
    rvec<int32_t> center_perm, center_row_ptr;
    rvec<int8_t> off_i8;     // [n_edges*3] integer images of dst (|img|<=127)
    int64_t n_owned = 0, n_owned_bonds = 0;
};

// stable counting sort of [0..m) by key[i] in [0..nkeys); emits perm and
// CSR row_ptr[nkeys+1].  Parallel two-pass with per-thread count matrices
// for large m (thread-count-independent output: chunk t's elements of
// bucket k land at base[k] + sum of earlier chunks' counts — stable).
template <class K>
void counting_csr(const K* key, int64_t m, int64_t nkeys,
                  rvec<int32_t>& perm, rvec<int32_t>& rptr) {
    perm.resize(m);
    rptr.resize(nkeys + 1);
    const int T = (m > 2'000'000) ? std::min(omp_get_max_threads(), 16) : 1;
    if (T == 1) {
        std::vector<int64_t> cnt(nkeys + 1, 0);
        for (int64_t i = 0; i < m; ++i) ++cnt[key[i] + 1];
        for (int64_t k = 0; k < nkeys; ++k) cnt[k + 1] += cnt[k];
        for (int64_t k = 0; k <= nkeys; ++k) rptr[k] = (int32_t)cnt[k];
        std::vector<int64_t> fill(nkeys, 0);
        for (int64_t i = 0; i < m; ++i)
            perm[cnt[key[i]] + fill[key[i]]++] = (int32_t)i;
        return;
    }
    rvec<int64_t> tcnt;
    tcnt.resize((int64_t)T * nkeys);
#pragma omp parallel num_threads(T)
    {
        const int t = omp_get_thread_num();
        int64_t* c = &tcnt[(int64_t)t * nkeys];
        std::memset(c, 0, nkeys * sizeof(int64_t));
        const int64_t lo = m * t / T, hi = m * (t + 1) / T;
        for (int64_t i = lo; i < hi; ++i) ++c[key[i]];
    }
    // exclusive scan over (bucket-major, thread-minor): c[t][k] becomes the
    // start offset for chunk t's elements of bucket k.  Blocked so the
    // strided (t, k) accesses stay cache-resident.
    int64_t run = 0;
    rptr[0] = 0;
    constexpr int64_t BL = 8192;
    std::vector<int64_t> blk((int64_t)T * BL);
    for (int64_t kb = 0; kb < nkeys; kb += BL) {
        const int64_t be = std::min(nkeys, kb + BL), bw = be - kb;
        for (int t = 0; t < T; ++t)
            std::memcpy(&blk[t * bw], &tcnt[(int64_t)t * nkeys + kb],
                        bw * sizeof(int64_t));
        for (int64_t k = 0; k < bw; ++k) {
            for (int t = 0; t < T; ++t) {
                const int64_t v = blk[t * bw + k];
                blk[t * bw + k] = run;
                run += v;
            }
            rptr[kb + k + 1] = (int32_t)run;
        }
        for (int t = 0; t < T; ++t)
            std::memcpy(&tcnt[(int64_t)t * nkeys + kb], &blk[t * bw],
                        bw * sizeof(int64_t));
    }
#pragma omp parallel num_threads(T)
    {
        const int t = omp_get_thread_num();
        int64_t* c = &tcnt[(int64_t)t * nkeys];
        const int64_t lo = m * t / T, hi = m * (t + 1) / T;
        for (int64_t i = lo; i < hi; ++i)
            perm[c[key[i]]++] = (int32_t)i;
    }
}

}  // namespace

struct dm_graph {
    int64_t n_atoms = 0, n_edges = 0, n_within = 0;
    int32_t P = 0;
    rvec<int64_t> src, dst, within;
    rvec<double> offsets, dist;
    std::vector<Part> parts;
};

namespace {

// ---------------------------------------------------------------------------
// neighbor list (contract of fpis.c:418-901)
//
// `frac` holds n PACKED (possibly subset) atoms; `sel` (nullable) maps the
// packed index to the GLOBAL atom id.  Edges are emitted center-major with
// each center's neighbors in CANONICAL (global dst id, image) order, so a
// focused slab build and a full build produce identical relative orders —
// the property the cross-rank halo slice alignment relies on.
// ---------------------------------------------------------------------------
int build_neighbor_list(dm_graph* g, const double* frac, const double* lat,
                        const int64_t* pbc, int64_t n, double r, double bond_r,
                        double tol, int nthreads, const int64_t* sel) {
    const double r2 = r * r, bond_r2 = bond_r * bond_r;
    double t_phase_ = omp_get_wtime();

    std::vector<double> cart(3 * n);
#pragma omp parallel for num_threads(nthreads) schedule(static)
    for (int64_t i = 0; i < n; ++i) {
        const double u = frac[3 * i], v = frac[3 * i + 1], w = frac[3 * i + 2];
        for (int k = 0; k < 3; ++k)
            cart[3 * i + k] = u * lat[0 + k] + v * lat[3 + k] + w * lat[6 + k];
    }

    // bounding box of the original cloud, padded by r+tol (fpis.c:480-487)
    double bmin[3] = {1e300, 1e300, 1e300}, bmax[3] = {-1e300, -1e300, -1e300};
    for (int64_t i = 0; i < n; ++i)
        for (int k = 0; k < 3; ++k) {
            bmin[k] = std::min(bmin[k], cart[3 * i + k]);
            bmax[k] = std::max(bmax[k], cart[3 * i + k]);
        }
    double vmin[3], vmax[3];
    for (int k = 0; k < 3; ++k) { vmin[k] = bmin[k] - r - tol; vmax[k] = bmax[k] + r + tol; }

    // image ranges from the reciprocal-lattice bound (fpis.c:255-262)
    // inv = inverse(lat); recip row k = 2*pi * inv^T row k = 2*pi * inv col k
    double det = lat[0] * (lat[4] * lat[8] - lat[5] * lat[7])
               - lat[1] * (lat[3] * lat[8] - lat[5] * lat[6])
               + lat[2] * (lat[3] * lat[7] - lat[4] * lat[6]);
    if (std::abs(det) < 1e-12) { set_err("singular lattice"); return -3; }
    double inv[9];
    inv[0] = (lat[4] * lat[8] - lat[5] * lat[7]) / det;
    inv[1] = (lat[2] * lat[7] - lat[1] * lat[8]) / det;
    inv[2] = (lat[1] * lat[5] - lat[2] * lat[4]) / det;
    inv[3] = (lat[5] * lat[6] - lat[3] * lat[8]) / det;
    inv[4] = (lat[0] * lat[8] - lat[2] * lat[6]) / det;
    inv[5] = (lat[2] * lat[3] - lat[0] * lat[5]) / det;
    inv[6] = (lat[3] * lat[7] - lat[4] * lat[6]) / det;
    inv[7] = (lat[1] * lat[6] - lat[0] * lat[7]) / det;
    inv[8] = (lat[0] * lat[4] - lat[1] * lat[3]) / det;
    int nmax[3];
    for (int k = 0; k < 3; ++k) {
        // |recip_k| = 2*pi*|inv column k|
        double cx = inv[k], cy = inv[3 + k], cz = inv[6 + k];
        double rl = std::sqrt(cx * cx + cy * cy + cz * cz);
        nmax[k] = pbc[k] ? (int)std::ceil((r + 0.15) * rl) : 0;
    }

    // expanded points: every atom in every candidate image, kept if inside
    // the padded bbox (count+fill, deterministic order: image-major, atom)
    struct XP { double x, y, z; int64_t orig; int16_t ia, ib, ic; };
    std::vector<XP> xp;
    {
        std::vector<std::array<int, 3>> imgs;
        for (int a = -nmax[0]; a <= nmax[0]; ++a)
            for (int b = -nmax[1]; b <= nmax[1]; ++b)
                for (int c = -nmax[2]; c <= nmax[2]; ++c)
                    imgs.push_back({a, b, c});
        const int64_t M = (int64_t)imgs.size();
        std::vector<int64_t> cnt(M + 1, 0);
#pragma omp parallel for num_threads(nthreads) schedule(dynamic, 1)
        for (int64_t m = 0; m < M; ++m) {
            const auto& im = imgs[m];
            double sx = im[0] * lat[0] + im[1] * lat[3] + im[2] * lat[6];
            double sy = im[0] * lat[1] + im[1] * lat[4] + im[2] * lat[7];
            double sz = im[0] * lat[2] + im[1] * lat[5] + im[2] * lat[8];
            int64_t c = 0;
            for (int64_t i = 0; i < n; ++i) {
                double x = cart[3 * i] + sx, y = cart[3 * i + 1] + sy, z = cart[3 * i + 2] + sz;
                if (x >= vmin[0] && x <= vmax[0] && y >= vmin[1] && y <= vmax[1] &&
                    z >= vmin[2] && z <= vmax[2]) ++c;
            }
            cnt[m + 1] = c;
        }
        for (int64_t m = 0; m < M; ++m) cnt[m + 1] += cnt[m];
        xp.resize(cnt[M]);
#pragma omp parallel for num_threads(nthreads) schedule(dynamic, 1)
        for (int64_t m = 0; m < M; ++m) {
            const auto& im = imgs[m];
            double sx = im[0] * lat[0] + im[1] * lat[3] + im[2] * lat[6];
            double sy = im[0] * lat[1] + im[1] * lat[4] + im[2] * lat[7];
            double sz = im[0] * lat[2] + im[1] * lat[5] + im[2] * lat[8];
            int64_t o = cnt[m];
            for (int64_t i = 0; i < n; ++i) {
                double x = cart[3 * i] + sx, y = cart[3 * i + 1] + sy, z = cart[3 * i + 2] + sz;
                if (x >= vmin[0] && x <= vmax[0] && y >= vmin[1] && y <= vmax[1] &&
                    z >= vmin[2] && z <= vmax[2])
                    xp[o++] = XP{x, y, z, i, (int16_t)im[0], (int16_t)im[1], (int16_t)im[2]};
            }
        }
    }

    DM_T("nl:expand");
    // linked-cell grid over the padded bbox, cell size >= r
    const int64_t nx = std::max<int64_t>(1, (int64_t)((vmax[0] - vmin[0]) / r));
    const int64_t ny = std::max<int64_t>(1, (int64_t)((vmax[1] - vmin[1]) / r));
    const int64_t nz = std::max<int64_t>(1, (int64_t)((vmax[2] - vmin[2]) / r));
    const double cwx = (vmax[0] - vmin[0]) / nx + 1e-12;
    const double cwy = (vmax[1] - vmin[1]) / ny + 1e-12;
    const double cwz = (vmax[2] - vmin[2]) / nz + 1e-12;
    const int64_t ncell = nx * ny * nz;
    auto cell_of = [&](double x, double y, double z) -> int64_t {
        int64_t cx = std::min<int64_t>(nx - 1, std::max<int64_t>(0, (int64_t)((x - vmin[0]) / cwx)));
        int64_t cy = std::min<int64_t>(ny - 1, std::max<int64_t>(0, (int64_t)((y - vmin[1]) / cwy)));
        int64_t cz = std::min<int64_t>(nz - 1, std::max<int64_t>(0, (int64_t)((z - vmin[2]) / cwz)));
        return (cx * ny + cy) * nz + cz;
    };
    const int64_t nxp = (int64_t)xp.size();
    std::vector<int64_t> cell_cnt(ncell + 1, 0), pt_cell(nxp), cell_fill(ncell, 0);
    for (int64_t j = 0; j < nxp; ++j) {
        pt_cell[j] = cell_of(xp[j].x, xp[j].y, xp[j].z);
        ++cell_cnt[pt_cell[j] + 1];
    }
    for (int64_t c = 0; c < ncell; ++c) cell_cnt[c + 1] += cell_cnt[c];
    std::vector<int32_t> cell_pts(nxp);
    for (int64_t j = 0; j < nxp; ++j)
        cell_pts[cell_cnt[pt_cell[j]] + cell_fill[pt_cell[j]]++] = (int32_t)j;

    DM_T("nl:bin");
    // per-center count + fill (deterministic center-major order)
    std::vector<int64_t> ecnt(n + 1, 0), wcnt(n + 1, 0);
    auto visit = [&](int64_t i, auto&& emit) {
        const double xi = cart[3 * i], yi = cart[3 * i + 1], zi = cart[3 * i + 2];
        int64_t cx = std::min<int64_t>(nx - 1, std::max<int64_t>(0, (int64_t)((xi - vmin[0]) / cwx)));
        int64_t cy = std::min<int64_t>(ny - 1, std::max<int64_t>(0, (int64_t)((yi - vmin[1]) / cwy)));
        int64_t cz = std::min<int64_t>(nz - 1, std::max<int64_t>(0, (int64_t)((zi - vmin[2]) / cwz)));
        for (int64_t ax = std::max<int64_t>(0, cx - 1); ax <= std::min(nx - 1, cx + 1); ++ax)
            for (int64_t ay = std::max<int64_t>(0, cy - 1); ay <= std::min(ny - 1, cy + 1); ++ay)
                for (int64_t az = std::max<int64_t>(0, cz - 1); az <= std::min(nz - 1, cz + 1); ++az) {
                    const int64_t c = (ax * ny + ay) * nz + az;
                    for (int64_t q = cell_cnt[c]; q < cell_cnt[c + 1]; ++q) {
                        const XP& p = xp[cell_pts[q]];
                        if (p.orig == i) continue;               // fpis.c:833
                        const double dx = p.x - xi, dy = p.y - yi, dz = p.z - zi;
                        const double d2 = dx * dx + dy * dy + dz * dz;
                        if (d2 < r2 + tol && d2 > tol) emit(p, d2);
                    }
                }
    };
#pragma omp parallel for num_threads(nthreads) schedule(dynamic, 256)
    for (int64_t i = 0; i < n; ++i) {
        int64_t e = 0, w = 0;
        visit(i, [&](const XP&, double d2) { ++e; if (d2 < bond_r2 + tol) ++w; });
        ecnt[i + 1] = e; wcnt[i + 1] = w;
    }
    DM_T("nl:count");
    for (int64_t i = 0; i < n; ++i) { ecnt[i + 1] += ecnt[i]; wcnt[i + 1] += wcnt[i]; }

    const int64_t E = ecnt[n], W = wcnt[n];
    g->n_edges = E; g->n_within = W;
    g->src.resize(E); g->dst.resize(E); g->offsets.resize(3 * E); g->dist.resize(E);
    g->within.resize(W);
    struct ERec { int64_t dst, key; int16_t ia, ib, ic; double d2; };
#pragma omp parallel num_threads(nthreads)
    {
        std::vector<ERec> buf;
        buf.reserve(512);
#pragma omp for schedule(dynamic, 256)
        for (int64_t i = 0; i < n; ++i) {
            buf.clear();
            visit(i, [&](const XP& p, double d2) {
                buf.push_back(ERec{p.orig, sel ? sel[p.orig] : p.orig,
                                   p.ia, p.ib, p.ic, d2});
            });
            std::sort(buf.begin(), buf.end(), [](const ERec& a, const ERec& b) {
                if (a.key != b.key) return a.key < b.key;
                if (a.ia != b.ia) return a.ia < b.ia;
                if (a.ib != b.ib) return a.ib < b.ib;
                return a.ic < b.ic;
            });
            int64_t e = ecnt[i], w = wcnt[i];
            for (const ERec& p : buf) {
                g->src[e] = i; g->dst[e] = p.dst;
                g->offsets[3 * e] = p.ia; g->offsets[3 * e + 1] = p.ib;
                g->offsets[3 * e + 2] = p.ic;
                g->dist[e] = std::sqrt(p.d2);
                if (p.d2 < bond_r2 + tol) g->within[w++] = e;
                ++e;
            }
        }
    }
    DM_T("nl:fill");
    return 0;
}

// ---------------------------------------------------------------------------
// partitioner (contract of utils.c:26-931)
// ---------------------------------------------------------------------------
struct Walls {
    int dim = 0;
    double fmin = 0, fmax = 0;
    std::vector<double> w;
    inline int which(const double* fc) const {                  // utils.c:1312-1322
        const double x = fc[dim];
        for (size_t i = 0; i < w.size(); ++i)
            if (x < w[i]) return (int)i;
        return (int)w.size();
    }
};

// partition rule over the FULL structure (utils.c:1370-1456 + width check
// utils.c:1512-1529).  Always global: a focused slab build must use the
// same walls as every other rank.
int compute_walls(const double* frac, const double* lat, int64_t n, int P,
                  double r, double bond_r, bool use_bond_graph, Walls& walls) {
    if (P <= 1) return 0;
    // QUIRK preserved: the reference picks the partition dimension from
    // "cartesian" coordinates computed as cart = L . frac with lattice
    // ROWS as the matrix rows (fast.c:16-27 fractional_to_cartesian),
    // i.e. the TRANSPOSE of the convention FPIS/ase use (cart = frac @ L).
    double cmin[3] = {1e300, 1e300, 1e300}, cmax[3] = {-1e300, -1e300, -1e300};
    for (int64_t i = 0; i < n; ++i) {
        const double u = frac[3 * i], v = frac[3 * i + 1], w = frac[3 * i + 2];
        double c[3];
        for (int k = 0; k < 3; ++k)
            c[k] = u * lat[3 * k] + v * lat[3 * k + 1] + w * lat[3 * k + 2];
        for (int k = 0; k < 3; ++k) { cmin[k] = std::min(cmin[k], c[k]); cmax[k] = std::max(cmax[k], c[k]); }
    }
    int dim = 0;
    for (int k = 1; k < 3; ++k)
        if (cmax[k] - cmin[k] > cmax[dim] - cmin[dim]) dim = k;
    double fmin = frac[dim], fmax = frac[dim];
    for (int64_t i = 1; i < n; ++i) {
        fmin = std::min(fmin, frac[3 * i + dim]);
        fmax = std::max(fmax, frac[3 * i + dim]);
    }
    const double flen = fmax - fmin;
    walls.dim = dim;
    walls.fmin = fmin;
    walls.fmax = fmax;
    walls.w.resize(P - 1);
    for (int i = 1; i < P; ++i)
        walls.w[i - 1] = i * (flen / P) + WALL_EPSILON + fmin;   // utils.c:1433-1435
    bool coll = true;                                            // utils.c:1440-1455
    while (coll) {
        coll = false;
        for (int wi = 0; wi < P - 1; ++wi)
            for (int64_t i = 0; i < n; ++i)
                if (frac[3 * i + dim] == walls.w[wi]) {
                    coll = true;
                    std::fprintf(stderr, "Collision b/w atom and partition wall, moving wall.\n");
                    walls.w[wi] += WALL_EPSILON;
                }
    }
    const double lv[3] = {lat[dim], lat[dim + 3], lat[dim + 6]};
    const double vnorm = std::sqrt(lv[0] * lv[0] + lv[1] * lv[1] + lv[2] * lv[2]);
    const double width = walls.w[0] * vnorm;
    if (use_bond_graph && width <= 2 * (r + bond_r)) {
        set_err("Partition walls are too close together: width " +
                std::to_string(width) + " <= 2*(cutoff+bond_cutoff)");
        return -4;
    }
    if (!use_bond_graph && width <= 2 * r) {
        set_err("Partition walls are too close together: width " +
                std::to_string(width) + " <= 2*cutoff");
        return -4;
    }
    return 0;
}

// `frac` holds n packed (possibly subset) atoms; `sel` maps packed -> global
// atom id (nullptr = identity).  With focus >= 0 only that partition's
// outputs are materialized (the others stay empty).
int build_partitions(dm_graph* g, const double* frac, const int64_t* sel,
                     int64_t n, const Walls& walls,
                     double r, double bond_r, int P, int nthreads,
                     bool use_bond_graph, int focus) {
    const int64_t E = g->n_edges;
    double t_phase_ = omp_get_wtime();
    g->P = P;
    g->parts.resize(P);

    // home partition per node
    std::vector<int32_t> home(n);
#pragma omp parallel for num_threads(nthreads) schedule(static)
    for (int64_t i = 0; i < n; ++i) home[i] = walls.which(&frac[3 * i]);

    // to_part: src of any cross-partition edge is sent to dst's partition
    // (utils.c:1189-1253); same-value races benign, conflicts warned below
    std::vector<int32_t> to_part(n, -1);
    std::atomic<int> conflicts{0};
#pragma omp parallel for num_threads(nthreads) schedule(static)
    for (int64_t e = 0; e < E; ++e) {
        const int ps = home[g->src[e]], pd = home[g->dst[e]];
        if (ps != pd) {
            int32_t prev = to_part[g->src[e]];
            if (prev != -1 && prev != pd) conflicts.fetch_add(1, std::memory_order_relaxed);
            to_part[g->src[e]] = pd;
        }
    }
    if (conflicts.load() > 0)
        std::fprintf(stderr,
                     "WARNING: %d border nodes straddle multiple partitions "
                     "(slab width < cutoff) — results may be incorrect "
                     "(reference utils.c:1243-1251 keeps last assignment).\n",
                     conflicts.load());

    // --- node buckets per partition, global order preserved (utils.c:1272-1297)
    // buckets hold PACKED indices; packed order == ascending global id
    // (sel is ascending), so region content order is rank-invariant — the
    // property the cross-rank halo slices rely on.
    std::vector<std::vector<int64_t>> pure(P);
    std::vector<std::vector<std::vector<int64_t>>> to_b(P), from_b(P);
    for (int p = 0; p < P; ++p) { to_b[p].resize(P); from_b[p].resize(P); }
    for (int64_t i = 0; i < n; ++i) {
        const int hp = home[i], tp = to_part[i];
        if (tp == -1) pure[hp].push_back(i);
        else { to_b[hp][tp].push_back(i); from_b[tp][hp].push_back(i); }
    }

    // markers + global_ids (utils.c:1102-1154); global_ids hold PACKED
    // indices here and are translated through sel at the end
    for (int p = 0; p < P; ++p) {
        if (focus >= 0 && p != focus) continue;
        Part& pt = g->parts[p];
        pt.markers.reserve(2 * P + 1);
        pt.markers.push_back(0);
        pt.global_ids.assign(pure[p].begin(), pure[p].end());
        for (int q = 0; q < P; ++q) {
            pt.markers.push_back((int64_t)pt.global_ids.size());
            if (q != p)
                pt.global_ids.insert(pt.global_ids.end(), to_b[p][q].begin(), to_b[p][q].end());
        }
        pt.n_owned = (int64_t)pt.global_ids.size();
        for (int q = 0; q < P; ++q) {
            pt.markers.push_back((int64_t)pt.global_ids.size());
            if (q != p)
                pt.global_ids.insert(pt.global_ids.end(), from_b[p][q].begin(), from_b[p][q].end());
        }
    }

    // --- per-partition edge lists + bond graph, sequential over partitions
    std::vector<int64_t> g2l_node(n), g2l_edge;
    if (use_bond_graph) g2l_edge.resize(E);

    // per-edge owner partition
    std::vector<int32_t> eowner(E);
#pragma omp parallel for num_threads(nthreads) schedule(static)
    for (int64_t e = 0; e < E; ++e) eowner[e] = home[g->dst[e]];   // utils.c:206

    DM_T("pt:classify");
    for (int p = 0; p < P; ++p) {
        if (focus >= 0 && p != focus) continue;
        Part& pt = g->parts[p];
        std::fill(g2l_node.begin(), g2l_node.end(), (int64_t)-1);
        for (int64_t i = 0; i < (int64_t)pt.global_ids.size(); ++i)
            g2l_node[pt.global_ids[i]] = i;

        // local edges: gather in global-id order (count+fill over thread
        // chunks), then stable counting sort by dst_local -> DST-SORTED
        // edge arrays + CSR row_ptr (the scatter-add layout)
        {
            const int T = nthreads;
            const int64_t Nn = (int64_t)pt.global_ids.size();
            std::vector<int64_t> tcnt(T + 1, 0);
#pragma omp parallel num_threads(T)
            {
                const int t = omp_get_thread_num();
                const int64_t lo = E * t / T, hi = E * (t + 1) / T;
                int64_t c = 0;
                for (int64_t e = lo; e < hi; ++e) if (eowner[e] == p) ++c;
                tcnt[t + 1] = c;
            }
            for (int t = 0; t < T; ++t) tcnt[t + 1] += tcnt[t];
            const int64_t Ep = tcnt[T];
            rvec<int64_t> tsrc, tdst, tgid;
            tsrc.resize(Ep); tdst.resize(Ep); tgid.resize(Ep);
#pragma omp parallel num_threads(T)
            {
                const int t = omp_get_thread_num();
                const int64_t lo = E * t / T, hi = E * (t + 1) / T;
                int64_t o = tcnt[t];
                for (int64_t e = lo; e < hi; ++e)
                    if (eowner[e] == p) {
                        tsrc[o] = g2l_node[g->src[e]];
                        tdst[o] = g2l_node[g->dst[e]];
                        tgid[o] = e;
                        ++o;
                    }
            }
            rvec<int32_t> eperm;
            counting_csr(tdst.data(), Ep, Nn, eperm, pt.row_ptr);
            pt.src_local.resize(Ep); pt.dst_local.resize(Ep); pt.edge_gids.resize(Ep);
            pt.off_i8.resize(3 * Ep);
#pragma omp parallel for num_threads(T) schedule(static)
            for (int64_t i = 0; i < Ep; ++i) {
                const int64_t j = eperm[i];
                pt.src_local[i] = (int32_t)tsrc[j];
                pt.dst_local[i] = (int32_t)tdst[j];
                pt.edge_gids[i] = tgid[j];
                pt.off_i8[3 * i] = (int8_t)g->offsets[3 * tgid[j]];
                pt.off_i8[3 * i + 1] = (int8_t)g->offsets[3 * tgid[j] + 1];
                pt.off_i8[3 * i + 2] = (int8_t)g->offsets[3 * tgid[j] + 2];
                if (use_bond_graph) g2l_edge[tgid[j]] = i;
            }
            counting_csr(pt.src_local.data(), Ep, Nn, pt.src_perm, pt.src_row_ptr);
        }
        DM_T("pt:edges");

        if (!use_bond_graph) continue;

        // --- BDE classification over within edges (utils.c:497-653)
        struct BRec { int64_t gid; int64_t src_a, dst_a; bool needs; };
        std::vector<BRec> bpure;
        std::vector<std::vector<BRec>> bto(P), bfrom(P);
        for (int64_t wi = 0; wi < g->n_within; ++wi) {
            const int64_t e = g->within[wi];
            const int64_t d = g->dst[e], s = g->src[e];
            if (g2l_node[d] == -1) continue;
            if (to_part[d] == p) {
                bfrom[home[d]].push_back({e, s, d, false});        // utils.c:513-548
            } else if (to_part[d] != -1) {
                bto[to_part[d]].push_back({e, s, d, true});        // utils.c:550-596
            } else if (home[d] == p) {
                bpure.push_back({e, s, d, true});                  // utils.c:601-649
            }
        }

        // local ids [pure | to_0.. | from_0..] + line markers (utils.c:973-1031)
        std::vector<BRec> bdes;
        pt.line_markers.reserve(2 * P + 1);
        pt.line_markers.push_back(0);
        bdes.insert(bdes.end(), bpure.begin(), bpure.end());
        for (int q = 0; q < P; ++q) {
            pt.line_markers.push_back((int64_t)bdes.size());
            if (q != p) bdes.insert(bdes.end(), bto[q].begin(), bto[q].end());
        }
        pt.n_owned_bonds = (int64_t)bdes.size();
        for (int q = 0; q < P; ++q) {
            pt.line_markers.push_back((int64_t)bdes.size());
            if (q != p) bdes.insert(bdes.end(), bfrom[q].begin(), bfrom[q].end());
        }
        const int64_t B = (int64_t)bdes.size();
        pt.bde_edge_gids.resize(B);
        for (int64_t b = 0; b < B; ++b) pt.bde_edge_gids[b] = bdes[b].gid;

        // bond_mapping pairs: owned BDEs only (pure + to), local ids both
        // sides (utils.c:589-593, 639-642, 686-689)
        pt.map_de.reserve(pt.n_owned_bonds);
        pt.map_ude.reserve(pt.n_owned_bonds);
        for (int64_t b = 0; b < pt.n_owned_bonds; ++b) {
            pt.map_de.push_back(g2l_edge[bdes[b].gid]);
            pt.map_ude.push_back(b);
        }

        // adjacency: BDEs grouped by src atom (counting sort, stable)
        std::vector<int64_t> acnt(n + 1, 0);
        for (int64_t b = 0; b < B; ++b) ++acnt[bdes[b].src_a + 1];
        for (int64_t i = 0; i < n; ++i) acnt[i + 1] += acnt[i];
        std::vector<int64_t> by_src(B), afill(n, 0);
        for (int64_t b = 0; b < B; ++b)
            by_src[acnt[bdes[b].src_a] + afill[bdes[b].src_a]++] = b;

        // line graph (utils.c:702-751): e1 -> e2 when src(e2)==dst(e1),
        // e2 needs_in_line, and dst(e2)!=src(e1); center = src(e2)
        std::vector<int64_t> lcnt(B + 1, 0);
#pragma omp parallel for num_threads(nthreads) schedule(dynamic, 512)
        for (int64_t b1 = 0; b1 < B; ++b1) {
            const int64_t a2 = bdes[b1].dst_a;
            int64_t c = 0;
            for (int64_t q = acnt[a2]; q < acnt[a2 + 1]; ++q) {
                const BRec& e2 = bdes[by_src[q]];
                if (e2.needs && e2.dst_a != bdes[b1].src_a) ++c;
            }
            lcnt[b1 + 1] = c;
        }
        for (int64_t b = 0; b < B; ++b) lcnt[b + 1] += lcnt[b];
        const int64_t L = lcnt[B];
        rvec<int64_t> tls, tld, tlc;
        tls.resize(L); tld.resize(L); tlc.resize(L);
#pragma omp parallel for num_threads(nthreads) schedule(dynamic, 512)
        for (int64_t b1 = 0; b1 < B; ++b1) {
            const int64_t a2 = bdes[b1].dst_a;
            int64_t o = lcnt[b1];
            for (int64_t q = acnt[a2]; q < acnt[a2 + 1]; ++q) {
                const int64_t b2 = by_src[q];
                const BRec& e2 = bdes[b2];
                if (e2.needs && e2.dst_a != bdes[b1].src_a) {
                    tls[o] = b1;
                    tld[o] = b2;
                    tlc[o] = g2l_node[e2.src_a];                   // utils.c:733,753-760
                    ++o;
                }
            }
        }
        // l_dst-sorted line arrays + CSRs (scatter layouts for the kernels)
        rvec<int32_t> lperm;
        counting_csr(tld.data(), L, B, lperm, pt.line_row_ptr);
        pt.line_src.resize(L); pt.line_dst.resize(L); pt.line_center.resize(L);
#pragma omp parallel for num_threads(nthreads) schedule(static)
        for (int64_t i = 0; i < L; ++i) {
            const int64_t j = lperm[i];
            pt.line_src[i] = (int32_t)tls[j];
            pt.line_dst[i] = (int32_t)tld[j];
            pt.line_center[i] = (int32_t)tlc[j];
        }
        counting_csr(pt.line_src.data(), L, B, pt.line_src_perm,
                     pt.line_src_row_ptr);
        counting_csr(pt.line_center.data(), L, (int64_t)pt.global_ids.size(),
                     pt.center_perm, pt.center_row_ptr);
        DM_T("pt:bonds+lines");
    }
    // translate packed indices -> global atom ids
    if (sel) {
        for (int p = 0; p < P; ++p) {
            Part& pt = g->parts[p];
            for (auto& gid : pt.global_ids) gid = sel[gid];
        }
    }
    return 0;
}

}  // namespace

// ---------------------------------------------------------------------------
// C ABI
// ---------------------------------------------------------------------------
extern "C" {

int dm_graph_build_focus(const double* frac_coords, const double* lattice,
                         const int64_t* pbc, int64_t n_atoms, double cutoff,
                         double bond_cutoff, double tol,
                         int32_t num_partitions, int32_t num_threads,
                         int32_t use_bond_graph, int32_t focus,
                         dm_graph** out) {
    if (num_partitions <= 0) { set_err("num_partitions must be >= 1"); return -2; }
    if (num_threads <= 0) num_threads = 1;
    if (focus >= num_partitions) { set_err("focus >= num_partitions"); return -2; }
    auto* g = new dm_graph();
    g->n_atoms = n_atoms;

    Walls walls;
    int rc = compute_walls(frac_coords, lattice, n_atoms, num_partitions,
                           cutoff, bond_cutoff, use_bond_graph != 0, walls);
    if (rc != 0) { delete g; return rc; }

    // focused slab build: keep only atoms within (cutoff + bond_cutoff +
    // 1 A) of the focus partition's slab (fractional margin via the
    // inverse-lattice column norm), with a periodic wrap arm for the
    // first/last slabs.  Everything relevant to the focus partition —
    // its atoms, its ghosts (<= cutoff outside) and the src atoms of its
    // ghost BDEs (<= cutoff + bond_cutoff outside) — lies inside; the
    // slab-width check guarantees no atom interacts across a full slab.
    rvec<int64_t> sel;
    rvec<double> fsub;
    const double* f_use = frac_coords;
    int64_t n_use = n_atoms;
    const int64_t* sel_p = nullptr;
    if (focus >= 0 && num_partitions > 1) {
        const int d = walls.dim;
        // || column d of L^{-1} ||: bound on d(frac_d)/d(cart)
        double det = lattice[0] * (lattice[4] * lattice[8] - lattice[5] * lattice[7])
                   - lattice[1] * (lattice[3] * lattice[8] - lattice[5] * lattice[6])
                   + lattice[2] * (lattice[3] * lattice[7] - lattice[4] * lattice[6]);
        double ic0, ic1, ic2;
        if (d == 0) { ic0 = (lattice[4]*lattice[8]-lattice[5]*lattice[7]);
                      ic1 = (lattice[5]*lattice[6]-lattice[3]*lattice[8]);
                      ic2 = (lattice[3]*lattice[7]-lattice[4]*lattice[6]); }
        else if (d == 1) { ic0 = (lattice[2]*lattice[7]-lattice[1]*lattice[8]);
                           ic1 = (lattice[0]*lattice[8]-lattice[2]*lattice[6]);
                           ic2 = (lattice[1]*lattice[6]-lattice[0]*lattice[7]); }
        else { ic0 = (lattice[1]*lattice[5]-lattice[2]*lattice[4]);
               ic1 = (lattice[2]*lattice[3]-lattice[0]*lattice[5]);
               ic2 = (lattice[0]*lattice[4]-lattice[1]*lattice[3]); }
        const double colnorm = std::sqrt(ic0*ic0 + ic1*ic1 + ic2*ic2) / std::abs(det);
        const double m = (cutoff + bond_cutoff + 1.0) * colnorm;
        const double lo = (focus == 0 ? walls.fmin : walls.w[focus - 1]) - m;
        const double hi = (focus == num_partitions - 1 ? walls.fmax
                                                       : walls.w[focus]) + m;
        const bool wrap = pbc[d] != 0;
        sel.reserve(n_atoms / num_partitions * 2);
        for (int64_t i = 0; i < n_atoms; ++i) {
            const double f = frac_coords[3 * i + d];
            bool in = (f >= lo && f <= hi);
            if (!in && wrap) in = (f - 1.0 >= lo) || (f + 1.0 <= hi);
            if (in) sel.push_back(i);
        }
        const int64_t ns = (int64_t)sel.size();
        fsub.resize(3 * ns);
#pragma omp parallel for num_threads(num_threads) schedule(static)
        for (int64_t i = 0; i < ns; ++i) {
            fsub[3 * i] = frac_coords[3 * sel[i]];
            fsub[3 * i + 1] = frac_coords[3 * sel[i] + 1];
            fsub[3 * i + 2] = frac_coords[3 * sel[i] + 2];
        }
        f_use = fsub.data();
        n_use = ns;
        sel_p = sel.data();
    }

    rc = build_neighbor_list(g, f_use, lattice, pbc, n_use, cutoff,
                             bond_cutoff, tol, num_threads, sel_p);
    if (rc == 0)
        rc = build_partitions(g, f_use, sel_p, n_use, walls, cutoff,
                              bond_cutoff, num_partitions, num_threads,
                              use_bond_graph != 0,
                              num_partitions > 1 ? focus : -1);
    if (rc != 0) { delete g; return rc; }
    *out = g;
    return 0;
}

int dm_graph_build(const double* frac_coords, const double* lattice,
                   const int64_t* pbc, int64_t n_atoms, double cutoff,
                   double bond_cutoff, double tol, int32_t num_partitions,
                   int32_t num_threads, int32_t use_bond_graph,
                   dm_graph** out) {
    return dm_graph_build_focus(frac_coords, lattice, pbc, n_atoms, cutoff,
                                bond_cutoff, tol, num_partitions, num_threads,
                                use_bond_graph, -1, out);
}

int dm_graph_global_view(const dm_graph* g, dm_global_view* out) {
    if (!g || !out) return -1;
    out->n_atoms = g->n_atoms;
    out->n_edges = g->n_edges;
    out->n_within = g->n_within;
    out->num_partitions = g->P;
    out->src = g->src.data();
    out->dst = g->dst.data();
    out->offsets = g->offsets.data();
    out->dist = g->dist.data();
    out->within = g->within.data();
    return 0;
}

int dm_graph_partition_view(const dm_graph* g, int32_t partition,
                            dm_partition_view* out) {
    if (!g || !out || partition < 0 || partition >= g->P) return -1;
    const Part& pt = g->parts[partition];
    out->n_nodes = (int64_t)pt.global_ids.size();
    out->n_owned = pt.n_owned;
    out->n_edges = (int64_t)pt.src_local.size();
    out->n_bonds = (int64_t)pt.bde_edge_gids.size();
    out->n_owned_bonds = pt.n_owned_bonds;
    out->n_lines = (int64_t)pt.line_src.size();
    out->n_mapping = (int64_t)pt.map_de.size();
    out->markers = pt.markers.data();
    out->global_ids = pt.global_ids.data();
    out->src_local = pt.src_local.data();
    out->dst_local = pt.dst_local.data();
    out->edge_gids = pt.edge_gids.data();
    out->line_markers = pt.line_markers.empty() ? nullptr : pt.line_markers.data();
    out->line_src = pt.line_src.data();
    out->line_dst = pt.line_dst.data();
    out->line_center = pt.line_center.data();
    out->map_de = pt.map_de.data();
    out->map_ude = pt.map_ude.data();
    out->bde_edge_gids = pt.bde_edge_gids.data();
    out->row_ptr = pt.row_ptr.data();
    out->src_perm = pt.src_perm.data();
    out->src_row_ptr = pt.src_row_ptr.data();
    out->line_row_ptr = pt.line_row_ptr.empty() ? nullptr : pt.line_row_ptr.data();
    out->line_src_perm = pt.line_src_perm.data();
    out->line_src_row_ptr = pt.line_src_row_ptr.empty() ? nullptr : pt.line_src_row_ptr.data();
    out->center_perm = pt.center_perm.data();
    out->center_row_ptr = pt.center_row_ptr.empty() ? nullptr : pt.center_row_ptr.data();
    out->offsets_i8 = pt.off_i8.data();
    return 0;
}

void dm_graph_free(dm_graph* g) { delete g; }

const char* dm_last_error(void) { return g_err.c_str(); }

}  // extern "C"
