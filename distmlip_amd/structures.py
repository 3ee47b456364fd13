"""Synthetic structure generators (seeded) — the committed input generators
named by SURVEY.md §8(d).  Pure input synthesis: no oracle, no GPU.

All lattices are returned as 3x3 matrices whose ROWS are lattice vectors;
cart = frac @ lattice (matches fast.c:10-27 / dist.py:149).
"""
from __future__ import annotations

from dataclasses import dataclass

import numpy as np


@dataclass
class Structure:
    """Minimal structure container (the build's stand-in for ase.Atoms).

    frac_coords are WRAPPED fractional coordinates; species are integer
    element-type indices into the model's element table.
    """
    frac_coords: np.ndarray          # (N,3) float64, wrapped
    lattice: np.ndarray              # (3,3) float64, rows = lattice vectors
    species: np.ndarray              # (N,) int64 element-type indices
    pbc: np.ndarray                  # (3,) int64

    @property
    def num_atoms(self) -> int:
        return len(self.frac_coords)

    @property
    def cart_coords(self) -> np.ndarray:
        return self.frac_coords @ self.lattice

    @classmethod
    def from_ase(cls, atoms, element_to_index):
        """Adapter for ase.Atoms (ase is optional and not installed here)."""
        species = np.array([element_to_index[s] for s in atoms.get_chemical_symbols()],
                           dtype=np.int64)
        return cls(
            frac_coords=np.array(atoms.get_scaled_positions(wrap=True), dtype=np.float64),
            lattice=np.array(atoms.get_cell(), dtype=np.float64),
            species=species,
            pbc=atoms.get_pbc().astype(np.int64),
        )


def _jitter(frac, lattice, sigma, rng):
    """Cartesian Gaussian jitter (sigma in Angstrom), re-wrapped."""
    cart = frac @ lattice
    cart = cart + rng.normal(0.0, sigma, size=cart.shape)
    inv = np.linalg.inv(lattice)
    frac = cart @ inv
    return np.mod(np.mod(frac, 1.0), 1.0)


def diamond_si(reps: int, a: float = 5.43, jitter: float = 0.1, seed: int = 0,
               species_index: int = 0) -> Structure:
    """Diamond-cubic Si supercell, reps^3 conventional cells (8 atoms each).

    SURVEY §8(d) config #1: reps=5 -> 1,000 atoms; config #3: reps=50 -> 1.0M.
    """
    basis = np.array([
        [0.00, 0.00, 0.00], [0.50, 0.50, 0.00], [0.50, 0.00, 0.50], [0.00, 0.50, 0.50],
        [0.25, 0.25, 0.25], [0.75, 0.75, 0.25], [0.75, 0.25, 0.75], [0.25, 0.75, 0.75],
    ])
    return _cubic_supercell(basis, reps, a, jitter, seed, species_index)


def bcc_li(reps: int, a: float = 3.51, jitter: float = 0.1, seed: int = 0,
           species_index: int = 1) -> Structure:
    """BCC Li supercell (2 atoms / conventional cell).

    SURVEY §8(d) config #2: reps=37 -> 101,306 atoms (~100k).
    """
    basis = np.array([[0.0, 0.0, 0.0], [0.5, 0.5, 0.5]])
    return _cubic_supercell(basis, reps, a, jitter, seed, species_index)


def _cubic_supercell(basis, reps, a, jitter, seed, species_index) -> Structure:
    rng = np.random.default_rng(seed)
    reps3 = (reps, reps, reps) if np.isscalar(reps) else tuple(reps)
    rs = [np.arange(r) for r in reps3]
    cells = np.stack(np.meshgrid(*rs, indexing="ij"), axis=-1).reshape(-1, 3)
    frac = ((cells[:, None, :] + basis[None, :, :]) / np.array(reps3)).reshape(-1, 3)
    lattice = np.diag(a * np.array(reps3, dtype=float))
    if jitter > 0:
        frac = _jitter(frac, lattice, jitter, rng)
    n = len(frac)
    return Structure(
        frac_coords=np.ascontiguousarray(frac),
        lattice=lattice,
        species=np.full(n, species_index, dtype=np.int64),
        pbc=np.ones(3, dtype=np.int64),
    )


def random_cell(n_atoms: int, a: float = 12.0, n_species: int = 4, seed: int = 0,
                skew: float = 0.0) -> Structure:
    """Random atoms in a (possibly skewed) periodic cell — edge-case fodder."""
    rng = np.random.default_rng(seed)
    lattice = np.eye(3) * a
    if skew:
        lattice = lattice + rng.normal(0, skew * a, size=(3, 3)) * (1 - np.eye(3))
    frac = rng.random((n_atoms, 3))
    return Structure(
        frac_coords=np.ascontiguousarray(frac),
        lattice=lattice,
        species=rng.integers(0, n_species, n_atoms).astype(np.int64),
        pbc=np.ones(3, dtype=np.int64),
    )


def workload(name: str, n_gpus: int = 1, seed: int = 0) -> Structure:
    """Named bench workloads (BASELINE.json configs).

    "li100k"  : config #2 — BCC Li 37^3 x 2 = 101,306 atoms (per GPU; weak
                scaling grows the cube to keep ~100k atoms per rank).
    "si1m"    : config #3 — diamond Si 50^3 x 8 = 1,000,000 atoms (fixed).
    "si1k"    : config #1 — diamond Si 5^3 x 8 = 1,000 atoms.
    """
    if name == "si1k":
        return diamond_si(5, jitter=0.1, seed=seed)
    if name == "li100k":
        target = 101306 * n_gpus
        reps = int(round((target / 2) ** (1.0 / 3.0)))
        return bcc_li(reps, jitter=0.1, seed=seed)
    if name == "si1m":
        return diamond_si(50, jitter=0.1, seed=seed)
    if name == "si2m":
        return diamond_si(63, jitter=0.1, seed=seed)   # 2,000,376 atoms
    if name == "mace500k":
        # config #4 — MACE-MP-0 medium, 500k atoms fixed (8-GPU strong):
        # diamond Si 40^3 x 8 = 512,000 atoms, 3 random species
        s = diamond_si(40, jitter=0.1, seed=seed)
        rng = np.random.default_rng(seed + 77)
        s.species = rng.integers(0, 3, size=s.num_atoms).astype(np.int64)
        return s
    if name == "mace62k":
        # per-GPU unit of config #4 (500k/8), weak scaling with n_gpus
        target = 512000 // 8 * n_gpus
        reps = int(round((target / 8) ** (1.0 / 3.0)))
        s = diamond_si(reps, jitter=0.1, seed=seed)
        rng = np.random.default_rng(seed + 77)
        s.species = rng.integers(0, 3, size=s.num_atoms).astype(np.int64)
        return s
    if name == "uma2m":
        # config #5 — UMA bf16, 2M-atom amorphous structure (8-GPU):
        # jittered diamond-Si at amorphous-like disorder, 3 species
        s = diamond_si(63, jitter=0.35, seed=seed)    # 2,000,376 atoms
        rng = np.random.default_rng(seed + 99)
        s.species = rng.integers(0, 3, size=s.num_atoms).astype(np.int64)
        return s
    if name == "uma250k":
        # per-GPU unit of config #5 (2M/8), weak scaling with n_gpus
        target = 2000376 // 8 * n_gpus
        reps = int(round((target / 8) ** (1.0 / 3.0)))
        s = diamond_si(reps, jitter=0.35, seed=seed)
        rng = np.random.default_rng(seed + 99)
        s.species = rng.integers(0, 3, size=s.num_atoms).astype(np.int64)
        return s
    raise ValueError(f"unknown workload {name!r}")
