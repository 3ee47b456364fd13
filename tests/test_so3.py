"""Pins for the from-scratch SO(3) core (distmlip_amd/so3.py) against
INDEPENDENT sources: sympy's exact wigner_3j, scipy's complex spherical
harmonics, and numerically-checked equivariance/symmetry properties."""
import math

import numpy as np
import pytest
import torch

from distmlip_amd import so3


def test_wigner_3j_vs_sympy():
    from sympy.physics.wigner import wigner_3j as sympy_3j
    for j1 in range(4):
        for j2 in range(4):
            for j3 in range(abs(j1 - j2), min(j1 + j2, 4) + 1):
                for m1 in range(-j1, j1 + 1):
                    for m2 in range(-j2, j2 + 1):
                        m3 = -m1 - m2
                        if abs(m3) > j3:
                            continue
                        want = float(sympy_3j(j1, j2, j3, m1, m2, m3))
                        got = so3.wigner_3j(j1, j2, j3, m1, m2, m3)
                        assert abs(got - want) < 1e-13, (j1, j2, j3, m1, m2)


def test_real_sh_vs_scipy():
    """Hardcoded real SH == scipy complex SH through the real<->complex
    transform (same transform real_cg uses, so a convention mismatch
    between the two would fail here)."""
    try:
        from scipy.special import sph_harm_y

        def csh(l, m, theta_polar, phi_azim):
            return sph_harm_y(l, m, theta_polar, phi_azim)
    except ImportError:
        from scipy.special import sph_harm

        def csh(l, m, theta_polar, phi_azim):
            return sph_harm(m, l, phi_azim, theta_polar)

    rng = np.random.default_rng(3)
    v = rng.normal(size=(32, 3))
    v /= np.linalg.norm(v, axis=1, keepdims=True)
    theta = np.arccos(np.clip(v[:, 2], -1, 1))     # polar
    phi = np.arctan2(v[:, 1], v[:, 0])             # azimuth
    Y = so3.real_sh(torch.tensor(v), normalize=False).numpy()
    for l in range(so3.L_MAX + 1):
        U = so3._real_to_complex_U(l)
        Yc = np.stack([csh(l, mu, theta, phi)
                       for mu in range(-l, l + 1)], axis=1)  # [32, 2l+1]
        Yr = (Yc @ U.T).real * math.sqrt(4 * math.pi)  # component norm
        o = so3.L_OFF[l]
        assert np.allclose(Y[:, o:o + so3.L_DIMS[l]], Yr, atol=1e-12), l


def test_real_sh_component_norm_and_parity():
    rng = np.random.default_rng(5)
    v = torch.tensor(rng.normal(size=(64, 3)))
    Y = so3.real_sh(v)                            # normalized inside
    for l in range(so3.L_MAX + 1):
        o, d = so3.L_OFF[l], so3.L_DIMS[l]
        s = (Y[:, o:o + d] ** 2).sum(1)
        assert torch.allclose(s, torch.full_like(s, float(d)), atol=1e-10), l
    # parity: Y_l(-v) = (-1)^l Y_l(v)
    Ym = so3.real_sh(-v)
    for l in range(so3.L_MAX + 1):
        o, d = so3.L_OFF[l], so3.L_DIMS[l]
        assert torch.allclose(Ym[:, o:o + d],
                              ((-1.0) ** l) * Y[:, o:o + d], atol=1e-12)


def test_real_cg_equivariance_and_norm():
    R = so3.random_rotation(7)
    D = so3.wigner_D_from_sh(R)
    rng = np.random.default_rng(11)
    for l1 in range(4):
        for l2 in range(4):
            for l3 in range(abs(l1 - l2), min(l1 + l2, 3) + 1):
                C = so3.real_cg(l1, l2, l3)
                if np.linalg.norm(C) < 1e-9:
                    continue
                assert abs(np.linalg.norm(C) - 1.0) < 1e-12
                x = rng.normal(size=2 * l1 + 1)
                y = rng.normal(size=2 * l2 + 1)
                D1 = D[so3.L_OFF[l1]:so3.L_OFF[l1] + 2 * l1 + 1,
                       so3.L_OFF[l1]:so3.L_OFF[l1] + 2 * l1 + 1]
                D2 = D[so3.L_OFF[l2]:so3.L_OFF[l2] + 2 * l2 + 1,
                       so3.L_OFF[l2]:so3.L_OFF[l2] + 2 * l2 + 1]
                D3 = D[so3.L_OFF[l3]:so3.L_OFF[l3] + 2 * l3 + 1,
                       so3.L_OFF[l3]:so3.L_OFF[l3] + 2 * l3 + 1]
                lhs = np.einsum("ijk,i,j->k", C, D1 @ x, D2 @ y)
                rhs = D3 @ np.einsum("ijk,i,j->k", C, x, y)
                assert np.allclose(lhs, rhs, atol=1e-10), (l1, l2, l3)


def test_wigner_D_is_orthogonal_rep():
    Ra, Rb = so3.random_rotation(1), so3.random_rotation(2)
    Da, Db = so3.wigner_D_from_sh(Ra), so3.wigner_D_from_sh(Rb)
    Dab = so3.wigner_D_from_sh(Ra @ Rb)
    assert np.allclose(Da @ Da.T, np.eye(so3.SH_DIM), atol=1e-10)
    assert np.allclose(Da @ Db, Dab, atol=1e-9)    # homomorphism


@pytest.mark.parametrize("nu", [1, 2, 3])
@pytest.mark.parametrize("lo", [0, 1])
def test_symmetric_basis_properties(nu, lo):
    B = so3.symmetric_basis(nu, lo)               # [16]*nu + [do, P]
    P = B.shape[-1]
    assert P > 0
    flat = B.reshape(-1, P)
    # orthonormal paths
    assert np.allclose(flat.T @ flat, np.eye(P), atol=1e-10)
    # slot symmetry
    if nu >= 2:
        perm = (1, 0) + tuple(range(2, nu)) + (nu, nu + 1)
        assert np.allclose(B, B.transpose(perm), atol=1e-12)
    if nu == 3:
        assert np.allclose(B, B.transpose(0, 2, 1, 3, 4), atol=1e-12)
    # equivariance: contracting rotated inputs == rotating the output
    R = so3.random_rotation(9)
    D = so3.wigner_D_from_sh(R)
    do = 2 * lo + 1
    Dlo = D[so3.L_OFF[lo]:so3.L_OFF[lo] + do,
            so3.L_OFF[lo]:so3.L_OFF[lo] + do]
    rng = np.random.default_rng(13)
    xs = [rng.normal(size=so3.SH_DIM) for _ in range(nu)]
    for p in range(P):
        T = B[..., p]
        t_plain = T
        for x in xs:
            t_plain = np.tensordot(x, t_plain, axes=(0, 0))
        t_rot = T
        for x in xs:
            t_rot = np.tensordot(D @ x, t_rot, axes=(0, 0))
        assert np.allclose(t_rot, Dlo @ t_plain, atol=1e-9), (nu, lo, p)


def test_symmetric_basis_counts():
    """Path counts are determined by representation theory: the rank of
    the symmetrized coupling basis must equal the multiplicity of the
    O(3) irrep (lo, (-1)^lo) in Sym^nu(0e+1o+2e+3o), computed here
    INDEPENDENTLY by character quadrature over O(3)."""

    def chi_l(l, th):
        return np.sin((l + 0.5) * th) / np.sin(th / 2)

    def chiV(th, par):
        return sum((((-1.0) ** l) if par < 0 else 1.0) * chi_l(l, th)
                   for l in range(so3.L_MAX + 1))

    th = np.linspace(1e-6, math.pi, 20001)
    w = (1 - np.cos(th)) / math.pi              # Haar weight on classes

    def sym_mult(nu, lo):
        po = (-1) ** lo
        tot = 0.0
        for par in (+1, -1):
            c = chiV(th, par)
            if nu == 1:
                ch = c
            elif nu == 2:
                # x^2 of a rotoreflection is a rotation
                ch = (c ** 2 + chiV(2 * th, +1)) / 2
            else:
                ch = (c ** 3 + 3 * c * chiV(2 * th, +1)
                      + 2 * chiV(3 * th, par)) / 6
            tgt = (po if par < 0 else 1.0) * chi_l(lo, th)
            tot += 0.5 * np.trapezoid(ch * tgt * w, th)
        return round(float(tot))

    for nu in (1, 2, 3):
        for lo in (0, 1):
            got = so3.symmetric_basis(nu, lo).shape[-1]
            want = sym_mult(nu, lo)
            assert got == want, (nu, lo, got, want)
    # and the concrete numbers, so a quadrature regression is loud too
    assert so3.symmetric_basis(3, 0).shape[-1] == 8
    assert so3.symmetric_basis(3, 1).shape[-1] == 12
