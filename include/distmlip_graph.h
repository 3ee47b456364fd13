/* distmlip_graph.h — C-ABI of the native graph builder (CPU, OpenMP).
 *
 * This is the build's replacement for the reference's compiled graph layer:
 *   - neighbor list: replaces intra_parallel_find_points_in_spheres_c
 *     (reference distributed/fpis.c:418-901) — same emitted-edge contract
 *     (d^2 < r^2+tol, d^2 > tol, no self edges in any image; offsets are
 *     integer images of the dst atom relative to wrapped coordinates;
 *     within_bond_r flags d^2 < bond_r^2+tol).
 *   - slab partitioner + bond/line graph: replaces get_features
 *     (reference distributed/subgraph_creation_utils.c:26-931) — same
 *     partition rule (longest cartesian extent, fractional walls at
 *     min + i*len/P + 1e-10, collision nudge: utils.c:1370-1456), same
 *     node regions [pure | to_* | from_*] and marker layout
 *     (utils.c:1102-1154), same dst-owned edge rule (utils.c:206,235),
 *     same BDE classification and line-graph rule (utils.c:497-761).
 *   - the Python entry get_subgraphs_fast (reference
 *     subgraph_creation_fast.c:92-453) is mirrored one level up by
 *     distmlip_amd/capi.py over this ABI.
 *
 * Error codes mirror fast.c:205-212:  -2 bad num_partitions, -3 self edges
 * (cell smaller than cutoff), -4 partition walls too close
 * (utils.c:1512-1529).  Divergence: num_partitions == 1 is ALLOWED here
 * (single-GPU path; the reference refuses it, utils.c:48-52).
 *
 * Ownership: the handle owns every exported pointer; arrays stay valid
 * until dm_graph_free.  The caller never frees individual arrays.
 * Threading: build is internally OpenMP-parallel; the handle is immutable
 * after build and may be read from any thread.
 */
#ifndef DISTMLIP_GRAPH_H
#define DISTMLIP_GRAPH_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct dm_graph dm_graph;

/* Build the full graph + partition structure.
 * frac_coords: [n_atoms,3] row-major WRAPPED fractional coordinates.
 * lattice:     [3,3] row-major, ROWS are lattice vectors (cart = frac @ L).
 * pbc:         [3] flags.
 * Returns 0 on success, negative error code otherwise. */
int dm_graph_build(const double* frac_coords,
                   const double* lattice,
                   const int64_t* pbc,
                   int64_t n_atoms,
                   double cutoff,
                   double bond_cutoff,
                   double tol,
                   int32_t num_partitions,
                   int32_t num_threads,
                   int32_t use_bond_graph,
                   dm_graph** out);

/* Focused build (the SPMD per-rank fast path): identical walls/semantics,
 * but only atoms within (cutoff + bond_cutoff + 1 A) of partition `focus`'s
 * slab enter the neighbor list, and only that partition's outputs are
 * materialized — per-rank cost stays O(atoms per slab) instead of
 * O(total).  Region content orders are rank-invariant (ascending global
 * atom id; canonical per-center edge order), so marker halo slices align
 * across ranks built with different focuses.  With focus = -1 behaves
 * exactly like dm_graph_build.  NOTE: with focus >= 0 the global-view edge
 * arrays and per-partition edge_gids are SUBSET-local ids. */
int dm_graph_build_focus(const double* frac_coords,
                         const double* lattice,
                         const int64_t* pbc,
                         int64_t n_atoms,
                         double cutoff,
                         double bond_cutoff,
                         double tol,
                         int32_t num_partitions,
                         int32_t num_threads,
                         int32_t use_bond_graph,
                         int32_t focus,
                         dm_graph** out);

typedef struct {
    int64_t n_atoms;
    int64_t n_edges;            /* global directed edges */
    int64_t n_within;           /* edges within bond_cutoff */
    int32_t num_partitions;
    const int64_t* src;         /* [n_edges] global src atom (the center) */
    const int64_t* dst;         /* [n_edges] global dst atom */
    const double*  offsets;     /* [n_edges,3] integer image of dst */
    const double*  dist;        /* [n_edges] */
    const int64_t* within;      /* [n_within] edge indices with d <= bond_r */
} dm_global_view;

int dm_graph_global_view(const dm_graph* g, dm_global_view* out);

typedef struct {
    int64_t n_nodes;            /* local atoms incl ghosts */
    int64_t n_owned;            /* pure + to regions */
    int64_t n_edges;            /* local edges (dst owned here) */
    int64_t n_bonds;            /* local BDEs incl ghosts (0 if no bond graph) */
    int64_t n_owned_bonds;
    int64_t n_lines;            /* line-graph edges */
    int64_t n_mapping;          /* bond_mapping pairs (owned BDEs) */
    const int64_t* markers;        /* [2P+1] region starts (python appends total) */
    const int64_t* global_ids;     /* [n_nodes] local -> global atom id */
    const int32_t* src_local;      /* [n_edges] (int32: device-bound arrays
                                      are 32-bit so the host uploads them
                                      without a conversion pass) */
    const int32_t* dst_local;      /* [n_edges] */
    const int64_t* edge_gids;      /* [n_edges] local -> global edge id (L2G) */
    const int64_t* line_markers;   /* [2P+1] */
    const int32_t* line_src;       /* [n_lines] local BDE ids */
    const int32_t* line_dst;       /* [n_lines] */
    const int32_t* line_center;    /* [n_lines] LOCAL atom id of center */
    const int64_t* map_de;         /* [n_mapping] local edge id */
    const int64_t* map_ude;        /* [n_mapping] local BDE id */
    const int64_t* bde_edge_gids;  /* [n_bonds] global edge id per BDE,
                                      incl ghosts (build extension: lets a
                                      rank compute ghost bond geometry
                                      locally) */

    /* Build extensions for the HIP kernels: local edges are emitted
     * DST-SORTED (stable by global edge id within a dst), so the node
     * scatter-add is a contiguous segmented reduction; every other
     * scatter direction gets a permutation CSR (deterministic backward,
     * no atomics).  Line edges are likewise emitted l_dst-sorted. */
    const int32_t* row_ptr;            /* [n_nodes+1] CSR over dst_local   */
    const int32_t* src_perm;           /* [n_edges]  edge ids sorted by src */
    const int32_t* src_row_ptr;        /* [n_nodes+1]                      */
    const int32_t* line_row_ptr;       /* [n_bonds+1] CSR over line_dst    */
    const int32_t* line_src_perm;      /* [n_lines] line ids sorted by line_src */
    const int32_t* line_src_row_ptr;   /* [n_bonds+1]                      */
    const int32_t* center_perm;        /* [n_lines] line ids sorted by center atom */
    const int32_t* center_row_ptr;     /* [n_nodes+1]                      */
    const int8_t*  offsets_i8;         /* [n_edges,3] integer image of dst
                                          per LOCAL edge (|image| <= 127)  */
} dm_partition_view;

int dm_graph_partition_view(const dm_graph* g, int32_t partition,
                            dm_partition_view* out);

void dm_graph_free(dm_graph* g);

/* Last error message for this thread (valid until next failing call). */
const char* dm_last_error(void);

#ifdef __cplusplus
}
#endif
#endif /* DISTMLIP_GRAPH_H */
