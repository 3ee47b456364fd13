"""SO(3) equivariant math core for the MACE path — built from scratch.

The reference's MACE adapter delegates all equivariant arithmetic to
e3nn/mace-torch (imports at implementations/mace/models.py:1-38); neither
is installable here, so this module restates the required machinery from
the published definitions:

  * real spherical harmonics l = 0..3, e3nn-style 'component'
    normalization (sum_m Y_lm^2 = 2l+1 on the unit sphere), differentiable
    in torch (forces flow through them);
  * Wigner 3j symbols from the Racah closed form (exact factorial
    arithmetic via Python ints, evaluated to fp64);
  * real-basis Clebsch-Gordan coupling tensors C[l1,l2,l3] with unit
    Frobenius norm (the unitary real<->complex change of basis preserves
    the 3j normalization sum |3j|^2 = 1);
  * orthonormal bases of SYMMETRIZED equivariant maps V^{(x)nu} -> W for
    nu = 1,2,3 over the l=0..3 feature space V — the coupling structure of
    MACE's SymmetricContraction (correlation 3).  MACE's U matrices (from
    e3nn's ReducedTensorProducts) span the same space in a different
    orthonormal orientation; with learned weights the model families are
    identical, which is the relevant notion of architectural parity here
    (no pretrained checkpoint can be loaded without network access —
    DESIGN.md §MACE).

m ordering within each l is m = -l..l with the standard real-harmonic
correspondence (l=1 -> (y, z, x)).  Tests pin the numbers against
INDEPENDENT sources: sympy.physics.wigner.wigner_3j (exact rationals) and
scipy complex spherical harmonics through the real<->complex transform
(tests/test_so3.py).
"""
from __future__ import annotations

import math
from functools import lru_cache

import numpy as np
import torch

L_MAX = 3
L_DIMS = [2 * l + 1 for l in range(L_MAX + 1)]   # 1,3,5,7
SH_DIM = sum(L_DIMS)                              # 16
L_OFF = [0, 1, 4, 9]                              # offset of each l block


# ---------------------------------------------------------------------------
# real spherical harmonics (component normalization), torch, differentiable
# ---------------------------------------------------------------------------

def real_sh(vec: torch.Tensor, normalize: bool = True) -> torch.Tensor:
    """[..., 3] -> [..., 16] real SH l=0..3, m=-l..l per block, component
    normalization.  normalize=True first maps vec to the unit sphere
    (MACE: o3.SphericalHarmonics(..., normalize=True,
    normalization='component'))."""
    if normalize:
        vec = vec / torch.linalg.norm(vec, dim=-1, keepdim=True)
    x, y, z = vec[..., 0], vec[..., 1], vec[..., 2]
    x2, y2, z2 = x * x, y * y, z * z

    s3 = math.sqrt(3.0)
    s15 = math.sqrt(15.0)
    s5_2 = math.sqrt(5.0) / 2.0
    s15_2 = math.sqrt(15.0) / 2.0
    c3m3 = math.sqrt(70.0) / 4.0          # sqrt(4pi) * (1/4) sqrt(35/(2pi))
    c3m2 = math.sqrt(105.0)               # sqrt(4pi) * (1/2) sqrt(105/pi)
    c3m1 = math.sqrt(42.0) / 4.0          # sqrt(4pi) * (1/4) sqrt(21/(2pi))
    c30 = math.sqrt(7.0) / 2.0            # sqrt(4pi) * (1/4) sqrt(7/pi)

    out = [
        torch.ones_like(x),               # l=0
        s3 * y, s3 * z, s3 * x,           # l=1, m=-1,0,1
        s15 * x * y,                      # l=2, m=-2
        s15 * y * z,                      # m=-1
        s5_2 * (3 * z2 - 1),              # m=0
        s15 * x * z,                      # m=1
        s15_2 * (x2 - y2),                # m=2
        c3m3 * y * (3 * x2 - y2),         # l=3, m=-3
        c3m2 * x * y * z,                 # m=-2
        c3m1 * y * (5 * z2 - 1),          # m=-1
        c30 * (5 * z2 - 3) * z,           # m=0
        c3m1 * x * (5 * z2 - 1),          # m=1
        c3m2 * z * (x2 - y2) / 2.0,       # m=2
        c3m3 * x * (x2 - 3 * y2),         # m=3
    ]
    return torch.stack(out, dim=-1)


# ---------------------------------------------------------------------------
# Wigner 3j (Racah formula, exact integer factorials -> fp64)
# ---------------------------------------------------------------------------

@lru_cache(maxsize=None)
def wigner_3j(j1: int, j2: int, j3: int, m1: int, m2: int, m3: int) -> float:
    if m1 + m2 + m3 != 0:
        return 0.0
    if not (abs(j1 - j2) <= j3 <= j1 + j2):
        return 0.0
    if abs(m1) > j1 or abs(m2) > j2 or abs(m3) > j3:
        return 0.0
    f = math.factorial
    # triangle coefficient
    delta = (f(j1 + j2 - j3) * f(j1 - j2 + j3) * f(-j1 + j2 + j3)
             ) / f(j1 + j2 + j3 + 1)
    pref = delta * f(j1 + m1) * f(j1 - m1) * f(j2 + m2) * f(j2 - m2) \
        * f(j3 + m3) * f(j3 - m3)
    t_min = max(0, j2 - j3 - m1, j1 - j3 + m2)
    t_max = min(j1 + j2 - j3, j1 - m1, j2 + m2)
    s = 0.0
    for t in range(t_min, t_max + 1):
        denom = (f(t) * f(j3 - j2 + t + m1) * f(j3 - j1 + t - m2)
                 * f(j1 + j2 - j3 - t) * f(j1 - t - m1) * f(j2 - t + m2))
        s += (-1.0) ** t / denom
    return ((-1.0) ** (j1 - j2 - m3)) * math.sqrt(pref) * s


def _real_to_complex_U(l: int) -> np.ndarray:
    """U[m_real, mu_complex] with Y_real = U @ Y_complex (Condon-Shortley
    complex harmonics); rows ordered m = -l..l."""
    d = 2 * l + 1
    U = np.zeros((d, d), dtype=complex)
    rt2 = 1.0 / math.sqrt(2.0)
    for m in range(-l, l + 1):
        r = m + l
        if m == 0:
            U[r, l] = 1.0
        elif m > 0:
            U[r, m + l] = ((-1.0) ** m) * rt2
            U[r, -m + l] = rt2
        else:  # m < 0
            U[r, -m + l] = -1j * ((-1.0) ** m) * rt2
            U[r, m + l] = 1j * rt2
    return U


@lru_cache(maxsize=None)
def real_cg(l1: int, l2: int, l3: int) -> np.ndarray:
    """Real-basis coupling tensor C[d1, d2, d3], unit Frobenius norm.

    C = (U1 (x) U2 (x) U3) . wigner3j  up to a global phase chosen to make
    the tensor real (it is always purely real or purely imaginary)."""
    d1, d2, d3 = 2 * l1 + 1, 2 * l2 + 1, 2 * l3 + 1
    W = np.zeros((d1, d2, d3))
    for m1 in range(-l1, l1 + 1):
        for m2 in range(-l2, l2 + 1):
            m3 = -m1 - m2
            if abs(m3) <= l3:
                W[m1 + l1, m2 + l2, m3 + l3] = wigner_3j(l1, l2, l3,
                                                         m1, m2, m3)
    U1, U2, U3 = (_real_to_complex_U(l) for l in (l1, l2, l3))
    C = np.einsum("au,bv,cw,uvw->abc", U1, U2, U3, W.astype(complex))
    re, im = np.abs(C.real).max(), np.abs(C.imag).max()
    C = C.real if re >= im else C.imag
    nrm = np.linalg.norm(C)
    if nrm < 1e-12:
        return np.zeros((d1, d2, d3))
    C = C / nrm
    # deterministic sign: first nonzero entry positive
    flat = C.ravel()
    first = flat[np.abs(flat) > 1e-9][0]
    return C * np.sign(first)


def cg_nonzero(l1: int, l2: int, l3: int) -> bool:
    return abs(l1 - l2) <= l3 <= l1 + l2 and \
        np.linalg.norm(real_cg(l1, l2, l3)) > 1e-9


# ---------------------------------------------------------------------------
# symmetrized equivariant coupling bases (SymmetricContraction structure)
# ---------------------------------------------------------------------------

def _embed_pair(l1: int, l2: int, lo: int) -> np.ndarray:
    """CG(l1,l2->lo) embedded into the full V (x) V -> W_lo space,
    V = l=0..3 concatenated (dim 16)."""
    T = np.zeros((SH_DIM, SH_DIM, 2 * lo + 1))
    C = real_cg(l1, l2, lo)
    o1, o2 = L_OFF[l1], L_OFF[l2]
    T[o1:o1 + L_DIMS[l1], o2:o2 + L_DIMS[l2], :] = C
    return T


def _embed_triple(l1: int, l2: int, L: int, l3: int, lo: int) -> np.ndarray:
    """Coupling tree ((l1 (x) l2) -> L) (x) l3 -> lo embedded in
    V (x) V (x) V -> W_lo."""
    T = np.zeros((SH_DIM, SH_DIM, SH_DIM, 2 * lo + 1))
    C12 = real_cg(l1, l2, L)                       # [d1,d2,dL]
    C3 = real_cg(L, l3, lo)                        # [dL,d3,do]
    o1, o2, o3 = L_OFF[l1], L_OFF[l2], L_OFF[l3]
    blk = np.einsum("abL,Lco->abco", C12, C3)
    T[o1:o1 + L_DIMS[l1], o2:o2 + L_DIMS[l2], o3:o3 + L_DIMS[l3], :] = blk
    return T


def _orthonormal_span(tensors, tol: float = 1e-8):
    """Orthonormal basis (list) of the span of the given equal-shape
    tensors, via SVD on the flattened stack."""
    if not tensors:
        return []
    M = np.stack([t.ravel() for t in tensors])
    u, s, vt = np.linalg.svd(M, full_matrices=False)
    keep = s > tol * s[0] if s[0] > 0 else []
    basis = [vt[i].reshape(tensors[0].shape) for i in np.nonzero(keep)[0]]
    # deterministic sign
    out = []
    for b in basis:
        flat = b.ravel()
        first = flat[np.abs(flat) > 1e-9][0]
        out.append(b * np.sign(first))
    return out


@lru_cache(maxsize=None)
def symmetric_basis(nu: int, lo: int) -> np.ndarray:
    """Orthonormal basis of SYMMETRIC equivariant maps V^(x)nu -> W_lo,
    V = (l=0..3) 16-dim, returned as [16]*nu + [2lo+1, n_paths].

    nu=1: identity embeddings; nu=2: slot-symmetrized pair couplings;
    nu=3: slot-symmetrized coupling trees, orthonormalized (the span is
    basis-independent — see module docstring).

    O(3) parity filter: the feature space carries SH parity ((-1)^l per
    block) and MACE's targets are the PROPER irreps lo=0e / 1o (parity
    (-1)^lo), so only paths with (-1)^(sum l_i) == (-1)^lo couple —
    pseudo-tensor paths (0o, 1e, ...) are excluded exactly as e3nn's
    irreps arithmetic excludes them."""
    cands = []
    if nu == 1:
        if lo <= L_MAX:
            T = np.zeros((SH_DIM, 2 * lo + 1))
            o = L_OFF[lo]
            T[o:o + L_DIMS[lo], :] = np.eye(L_DIMS[lo])
            cands.append(T / np.linalg.norm(T))
    elif nu == 2:
        for l1 in range(L_MAX + 1):
            for l2 in range(L_MAX + 1):
                if (l1 + l2 + lo) % 2 == 0 and cg_nonzero(l1, l2, lo):
                    T = _embed_pair(l1, l2, lo)
                    Ts = 0.5 * (T + T.transpose(1, 0, 2))
                    if np.linalg.norm(Ts) > 1e-9:
                        cands.append(Ts)
    elif nu == 3:
        for l1 in range(L_MAX + 1):
            for l2 in range(L_MAX + 1):
                for L in range(abs(l1 - l2), min(l1 + l2, 2 * L_MAX) + 1):
                    if not cg_nonzero(l1, l2, L):
                        continue
                    for l3 in range(L_MAX + 1):
                        if (l1 + l2 + l3 + lo) % 2 != 0:
                            continue
                        if not cg_nonzero(L, l3, lo):
                            continue
                        T = _embed_triple(l1, l2, L, l3, lo)
                        Ts = sum(T.transpose(p + (3,)) for p in
                                 [(0, 1, 2), (0, 2, 1), (1, 0, 2),
                                  (1, 2, 0), (2, 0, 1), (2, 1, 0)]) / 6.0
                        if np.linalg.norm(Ts) > 1e-9:
                            cands.append(Ts)
    else:
        raise ValueError(nu)
    basis = _orthonormal_span(cands)
    if not basis:
        return np.zeros(tuple([SH_DIM] * nu) + (2 * lo + 1, 0))
    return np.stack(basis, axis=-1)


@lru_cache(maxsize=None)
def symmetric_basis_trees(nu: int, lo: int):
    """Tree factorization of `symmetric_basis(nu, lo)` for cheap
    contraction with the symmetric power x^(x)nu.

    Returns (trees, M):
      trees: list of coupling trees — nu=3: (l1, l2, L, l3); nu=2:
             (l1, l2); nu=1: (lo,)
      M:     [n_paths, n_trees] with  B_p = sum_t M[p,t] * Sym(T_t).
    Because x^(x)nu is symmetric, contracting with T_t equals contracting
    with Sym(T_t), so  phi_p(x) = sum_t M[p,t] * (T_t . x^nu)  — no dense
    [16]^nu tensors are ever materialized at model run time."""
    B = symmetric_basis(nu, lo)
    P = B.shape[-1]
    trees, sym_flat = [], []
    if nu == 1:
        if P:
            trees.append((lo,))
            # raw (unnormalized) tree tensor — the runtime contraction
            # computes plain x-slice / CG contractions, so S columns must
            # be the raw Sym(T_t), not renormalized
            T = np.zeros((SH_DIM, 2 * lo + 1))
            o = L_OFF[lo]
            T[o:o + L_DIMS[lo], :] = np.eye(L_DIMS[lo])
            sym_flat.append(T.ravel())
    elif nu == 2:
        for l1 in range(L_MAX + 1):
            for l2 in range(L_MAX + 1):
                if (l1 + l2 + lo) % 2 == 0 and cg_nonzero(l1, l2, lo):
                    T = _embed_pair(l1, l2, lo)
                    Ts = 0.5 * (T + T.transpose(1, 0, 2))
                    if np.linalg.norm(Ts) > 1e-9:
                        trees.append((l1, l2))
                        sym_flat.append(Ts.ravel())
    else:
        for l1 in range(L_MAX + 1):
            for l2 in range(L_MAX + 1):
                for L in range(abs(l1 - l2), min(l1 + l2, 2 * L_MAX) + 1):
                    if not cg_nonzero(l1, l2, L):
                        continue
                    for l3 in range(L_MAX + 1):
                        if (l1 + l2 + l3 + lo) % 2 != 0:
                            continue
                        if not cg_nonzero(L, l3, lo):
                            continue
                        T = _embed_triple(l1, l2, L, l3, lo)
                        Ts = sum(T.transpose(p + (3,)) for p in
                                 [(0, 1, 2), (0, 2, 1), (1, 0, 2),
                                  (1, 2, 0), (2, 0, 1), (2, 1, 0)]) / 6.0
                        if np.linalg.norm(Ts) > 1e-9:
                            trees.append((l1, l2, L, l3))
                            sym_flat.append(Ts.ravel())
    if not trees:
        return [], np.zeros((0, 0))
    S = np.stack(sym_flat, axis=1)                     # [dim, T]
    Bf = B.reshape(-1, P)                              # [dim, P]
    M, *_ = np.linalg.lstsq(S, Bf, rcond=None)         # S @ M = Bf
    resid = np.abs(S @ M - Bf).max()
    assert resid < 1e-9, f"tree decomposition failed: {resid}"
    return trees, M.T                                  # [P, T]


# ---------------------------------------------------------------------------
# Wigner D in this basis (tests + weight-free utilities)
# ---------------------------------------------------------------------------

def wigner_D_from_sh(R: np.ndarray) -> np.ndarray:
    """Block-diagonal [16,16] rotation matrix D(R) with
    real_sh(R v) = D(R) real_sh(v), solved per l from sampled unit
    vectors (uses only real_sh — no external convention)."""
    rng = np.random.default_rng(12345)
    V = rng.normal(size=(64, 3))
    V /= np.linalg.norm(V, axis=1, keepdims=True)
    Y = real_sh(torch.tensor(V), normalize=False).numpy()        # [64,16]
    YR = real_sh(torch.tensor(V @ R.T), normalize=False).numpy()
    D = np.zeros((SH_DIM, SH_DIM))
    for l in range(L_MAX + 1):
        o, d = L_OFF[l], L_DIMS[l]
        # YR_block = Y_block @ D_l^T  =>  D_l^T = lstsq(Y, YR)
        Dl_T, *_ = np.linalg.lstsq(Y[:, o:o + d], YR[:, o:o + d], rcond=None)
        D[o:o + d, o:o + d] = Dl_T.T
    return D


def random_rotation(seed: int = 0) -> np.ndarray:
    rng = np.random.default_rng(seed)
    A = rng.normal(size=(3, 3))
    Q, r = np.linalg.qr(A)
    Q *= np.sign(np.diag(r))
    if np.linalg.det(Q) < 0:
        Q[:, 0] = -Q[:, 0]
    return Q


# ---------------------------------------------------------------------------
# batched Wigner D from rotation matrices (round 2, UMA/eSCN path)
# ---------------------------------------------------------------------------

@lru_cache(maxsize=None)
def wigner_D_poly(l: int) -> np.ndarray:
    """T [9^l, dl*dl] with D_l(R).flatten() == kron(R,..,R).flatten() @ T
    for any ROTATION R: D_l entries are degree-l polynomials in R's
    entries, so T is solved once (lstsq over sampled rotations against
    wigner_D_from_sh).  Any solution agreeing on SO(3) has identical
    tangential derivatives along the manifold, so autograd gradients
    through batched evaluation are exact."""
    d = 2 * l + 1
    K = max(60, 3 * 9 ** l)
    rng = np.random.default_rng(100 + l)
    F = np.empty((K, 9 ** l))
    Y = np.empty((K, d * d))
    for k in range(K):
        A = rng.normal(size=(3, 3))
        Q, r = np.linalg.qr(A)
        Q *= np.sign(np.diag(r))
        if np.linalg.det(Q) < 0:
            Q[:, 0] = -Q[:, 0]
        f = np.array([1.0])
        for _ in range(l):
            f = np.kron(f, Q.ravel())
        F[k] = f
        D = wigner_D_from_sh(Q)
        o = L_OFF[l]
        Y[k] = D[o:o + d, o:o + d].ravel()
    T, *_ = np.linalg.lstsq(F, Y, rcond=None)
    resid = np.abs(F @ T - Y).max()
    assert resid < 1e-8, f"wigner_D_poly l={l}: residual {resid}"
    return T


def wigner_D_batch(R: torch.Tensor, l_max: int = L_MAX) -> torch.Tensor:
    """[..., 3, 3] rotations -> [..., S, S] block-diagonal Wigner D over
    l = 0..l_max (S = sum of 2l+1).  Differentiable in R; evaluation is
    kron-powers of R times the precomputed coefficient tensors."""
    S = sum(2 * ll + 1 for ll in range(l_max + 1))
    batch = R.shape[:-2]
    Rf = R.reshape(*batch, 9)
    D = R.new_zeros(*batch, S, S)
    D[..., 0, 0] = 1.0
    f = Rf
    off = 1
    for l in range(1, l_max + 1):
        d = 2 * l + 1
        T = torch.as_tensor(wigner_D_poly(l), dtype=R.dtype,
                            device=R.device)
        blk = (f @ T).reshape(*batch, d, d)
        D[..., off:off + d, off:off + d] = blk
        off += d
        if l < l_max:
            # next kron power: f_{l+1}[..., i*9+j] = f_l[..., i] * Rf[..., j]
            f = (f.unsqueeze(-1) * Rf.unsqueeze(-2)).reshape(*batch, -1)
    return D


def edge_align_rotation(vec: torch.Tensor) -> torch.Tensor:
    """[..., 3] -> [..., 3, 3] rotation R with R @ v_hat = z_hat,
    differentiable a.e. (the reference-axis pick is piecewise constant,
    like eSCN's perpendicular construction)."""
    v = vec / torch.linalg.norm(vec, dim=-1, keepdim=True)
    # reference axis: the canonical axis least aligned with v
    a = torch.zeros_like(v)
    idx = v.abs().argmin(dim=-1, keepdim=True)
    a.scatter_(-1, idx, 1.0)
    r1 = torch.cross(a, v, dim=-1)
    r1 = r1 / torch.linalg.norm(r1, dim=-1, keepdim=True)
    r2 = torch.cross(v, r1, dim=-1)                  # right-handed: r1 x r2 = v
    return torch.stack([r1, r2, v], dim=-2)
