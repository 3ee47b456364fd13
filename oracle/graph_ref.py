"""Independent numpy restatement of the reference's graph layer semantics.

This is the CPU oracle for the neighbor list: a brute-force O(N^2 * images)
periodic search restating the BEHAVIOR of the reference's FPIS
(`distributed/fpis.c:418-901`), written from the emitted-edge contract:

  * an edge (src=i, dst=j, offset=o, dist=d) exists iff
    d^2 = |cart(j) + o @ lattice - cart(i)|^2 satisfies
    d^2 < r^2 + tol  and  d^2 > tol  and  i != j      (fpis.c:833; self
    edges are excluded for EVERY image, i.e. an atom never bonds to its
    own periodic replica)
  * offsets are integer image vectors applied to the dst atom, relative to
    the WRAPPED dst position (fpis.c:838-840: the emitted offset folds the
    wrap correction of the dst atom in, and the Python layer always feeds
    wrapped fractional coordinates — dist.py:182, pes.py:71)
  * `within_bond_r` marks edges with d^2 < bond_r^2 + tol (fpis.c:843)

Edge ORDER is not part of the contract (the reference's order is its
cell-list traversal order); comparisons canonicalize on the sorted
(src, dst, offset) key.

Only tests may import this module (see oracle/__init__.py).
"""
from __future__ import annotations

import numpy as np


def brute_force_neighbors(frac_coords: np.ndarray,
                          lattice: np.ndarray,
                          pbc,
                          r: float,
                          bond_r: float = 0.0,
                          tol: float = 1e-8):
    """O(N^2) PBC neighbor search.

    Args:
        frac_coords: (N,3) WRAPPED fractional coordinates.
        lattice: (3,3) lattice matrix, ROWS are lattice vectors
                 (cart = frac @ lattice; fast.c:10-27).
        pbc: length-3 int/bool flags.
        r: neighbor cutoff (Angstrom).
        bond_r: three-body cutoff; edges with d^2 < bond_r^2 + tol are
                flagged in `within_bond_r`.
        tol: numerical tolerance (reference default 1e-8).

    Returns dict with src, dst (int64[E]), offsets (float64[E,3] integer
    image vectors of dst), dist (float64[E]), within_bond_r (int64 indices
    into the edge arrays).  Order: src-major, then dst, then image — a
    deterministic canonical order (NOT the reference's order).
    """
    frac = np.asarray(frac_coords, dtype=np.float64)
    lat = np.asarray(lattice, dtype=np.float64)
    pbc = np.asarray(pbc).astype(bool)
    n = len(frac)
    cart = frac @ lat

    # image range needed: enough images so that every point within r of the
    # cell is covered.  Use the reciprocal-lattice bound the reference uses
    # (fpis.c:261: maxr = ceil((r+0.15)*|b_i| / 2pi)).
    recip = 2 * np.pi * np.linalg.inv(lat).T
    nmax = np.ceil((r + 0.15) * np.linalg.norm(recip, axis=1) / (2 * np.pi)).astype(int)
    ranges = [range(-nmax[k], nmax[k] + 1) if pbc[k] else range(0, 1) for k in range(3)]

    images = np.array([[a, b, c] for a in ranges[0] for b in ranges[1] for c in ranges[2]],
                      dtype=np.float64)
    shifts = images @ lat  # (M,3)

    src_l, dst_l, off_l, d_l = [], [], [], []
    r2 = r * r
    for i in range(n):
        # displacement from i to every image of every j
        # d[j,m] = cart[j] + shifts[m] - cart[i]
        disp = cart[None, :, :] + shifts[:, None, :] - cart[i][None, None, :]  # (M,N,3)
        d2 = np.einsum("mjk,mjk->mj", disp, disp)
        m_idx, j_idx = np.nonzero((d2 < r2 + tol) & (d2 > tol))
        keep = j_idx != i  # no self edges in ANY image (fpis.c:833)
        m_idx, j_idx = m_idx[keep], j_idx[keep]
        order = np.lexsort((images[m_idx, 2], images[m_idx, 1], images[m_idx, 0], j_idx))
        m_idx, j_idx = m_idx[order], j_idx[order]
        src_l.append(np.full(len(j_idx), i, dtype=np.int64))
        dst_l.append(j_idx.astype(np.int64))
        off_l.append(images[m_idx])
        d_l.append(np.sqrt(d2[m_idx, j_idx]))

    src = np.concatenate(src_l) if src_l else np.zeros(0, np.int64)
    dst = np.concatenate(dst_l) if dst_l else np.zeros(0, np.int64)
    off = np.concatenate(off_l) if off_l else np.zeros((0, 3))
    dist = np.concatenate(d_l) if d_l else np.zeros(0)
    within = np.nonzero(dist * dist < bond_r * bond_r + tol)[0].astype(np.int64)
    return dict(src=src, dst=dst, offsets=off, dist=dist, within_bond_r=within)


def edge_key(src, dst, offsets):
    """Canonical sortable key array for edge-set comparison."""
    off = np.rint(np.asarray(offsets)).astype(np.int64)
    return np.stack([np.asarray(src, dtype=np.int64),
                     np.asarray(dst, dtype=np.int64),
                     off[:, 0], off[:, 1], off[:, 2]], axis=1)


def canonical_edge_order(src, dst, offsets):
    k = edge_key(src, dst, offsets)
    return np.lexsort((k[:, 4], k[:, 3], k[:, 2], k[:, 1], k[:, 0]))
