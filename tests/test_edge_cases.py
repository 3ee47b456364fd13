"""Edge cases: mixed/absent PBC, tiny systems, wall collisions, and
no-bond-graph builds — native builder vs reference module / brute force."""
import numpy as np
import pytest

from distmlip_amd import capi
from distmlip_amd.structures import Structure, diamond_si, random_cell
from oracle.graph_ref import brute_force_neighbors, canonical_edge_order, edge_key


def _assert_edges_match(ours, s, r=6.0, br=3.0):
    bf = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, r, br)
    k1, o1 = edge_key(ours[5], ours[6], ours[7]), \
        canonical_edge_order(ours[5], ours[6], ours[7])
    k2, o2 = edge_key(bf["src"], bf["dst"], bf["offsets"]), \
        canonical_edge_order(bf["src"], bf["dst"], bf["offsets"])
    assert k1.shape == k2.shape and (k1[o1] == k2[o2]).all()
    if len(k1):
        assert np.abs(np.asarray(ours[8])[o1] - bf["dist"][o2]).max() < 1e-12


@pytest.mark.parametrize("pbc", [[1, 1, 0], [0, 1, 1], [0, 0, 0]])
def test_mixed_pbc(pbc):
    s = random_cell(120, a=40.0, seed=11)
    s.pbc = np.array(pbc, dtype=np.int64)
    ours = capi.get_subgraphs_fast(s.cart_coords, 6.0, s.pbc, s.lattice, 2,
                                   3.0, 1e-8, 4, True, s.frac_coords)
    _assert_edges_match(ours, s)


def test_two_atoms():
    s = Structure(frac_coords=np.array([[0.4, 0.5, 0.5], [0.6, 0.5, 0.5]]),
                  lattice=np.eye(3) * 40.0,
                  species=np.zeros(2, dtype=np.int64),
                  pbc=np.ones(3, dtype=np.int64))
    ours = capi.get_subgraphs_fast(s.cart_coords, 6.0, s.pbc, s.lattice, 1,
                                   3.0, 1e-8, 2, True, s.frac_coords)
    # 16 A apart in a 40 A box: no edges at r=6
    assert len(ours[5]) == 0
    s2 = Structure(frac_coords=np.array([[0.47, 0.5, 0.5], [0.53, 0.5, 0.5]]),
                   lattice=np.eye(3) * 40.0,
                   species=np.zeros(2, dtype=np.int64),
                   pbc=np.ones(3, dtype=np.int64))
    ours2 = capi.get_subgraphs_fast(s2.cart_coords, 6.0, s2.pbc, s2.lattice, 1,
                                    3.0, 1e-8, 2, True, s2.frac_coords)
    assert len(ours2[5]) == 2          # both directions, 2.4 A apart
    assert len(ours2[11]) == 2         # within bond cutoff too


def test_wall_collision_nudge(ref_graph_backend):
    """Atom placed exactly at the wall coordinate: both builders nudge the
    wall (utils.c:1442-1455) and still produce matching region sizes."""
    s = random_cell(150, a=45.0, seed=13)
    # wall position for P=2 along the longest cart dim (cubic: dim picked
    # by extents; force dim 0 by stretching)
    s.lattice[0, 0] = 50.0
    frac = s.frac_coords.copy()
    fmin, fmax = frac[:, 0].min(), frac[:, 0].max()
    wall = (fmax - fmin) / 2 + 1e-10 + fmin
    frac[0, 0] = wall                   # exact collision
    s.frac_coords = frac
    ours = capi.get_subgraphs_fast(s.cart_coords, 6.0, s.pbc, s.lattice, 2,
                                   3.0, 1e-8, 2, True, s.frac_coords)
    ref = ref_graph_backend(s.cart_coords, 6.0, s.pbc, s.lattice, 2,
                            3.0, 1e-8, 2, True, s.frac_coords)
    for p in range(2):
        assert ours[2][p].tolist() == np.asarray(ref[2][p]).tolist()


def test_no_bond_graph_build():
    s = diamond_si((8, 2, 2), jitter=0.1, seed=4)
    out = capi.get_subgraphs_fast(s.cart_coords, 6.0, s.pbc, s.lattice, 2,
                                  3.0, 1e-8, 4, False, s.frac_coords)
    assert len(out[9][0]) == 0          # no line graph
    assert len(out[0][0]) + len(out[0][1]) == len(out[5])


def test_straddle_warning_does_not_crash(capfd):
    """Slab narrower than the cutoff (no bond graph, passes the width
    check marginally) — the reference warns and keeps the last assignment
    (utils.c:1243-1251); we must not crash and must produce a covering
    partition."""
    s = diamond_si((5, 1, 1), jitter=0.05, seed=1)   # 27.2 x 5.4 x 5.4 A
    with pytest.raises(RuntimeError):
        # width 13.6 <= 2*(6+3): hard error with bond graph (utils.c:1519)
        capi.get_subgraphs_fast(s.cart_coords, 6.0, s.pbc, s.lattice, 2,
                                3.0, 1e-8, 2, True, s.frac_coords)
    # without bond graph 13.6 > 2*6 passes
    out = capi.get_subgraphs_fast(s.cart_coords, 6.0, s.pbc, s.lattice, 2,
                                  3.0, 1e-8, 2, False, s.frac_coords)
    gids = np.sort(np.concatenate([
        np.asarray(out[4][p])[:out[2][p][3]] for p in range(2)]))
    assert (gids == np.arange(s.num_atoms)).all()


def test_engine_empty_bond_graph():
    """Engine E+F on a structure whose three-body set is EMPTY (simple
    cubic, nearest neighbor 4 A: inside the 6 A edge cutoff, outside the
    3 A bond cutoff) — the bond-graph branches must degrade cleanly and
    match the oracle."""
    import torch

    from distmlip_amd.model import CHGNetCore
    from distmlip_amd.runtime import SpmdEngine
    from distmlip_amd.structures import Structure
    from oracle.chgnet_ref import CpuRefOps, oracle_forward
    from oracle.graph_ref import brute_force_neighbors

    rng = np.random.default_rng(1)
    n = 4
    cells = np.stack(np.meshgrid(*[np.arange(n)] * 3, indexing="ij"),
                     -1).reshape(-1, 3)
    frac = (cells + 0.5) / n + rng.normal(0, 0.004, (n ** 3, 3))
    s = Structure(frac_coords=np.mod(frac, 1.0),
                  lattice=np.eye(3) * (4.0 * n),
                  species=np.zeros(n ** 3, dtype=np.int64),
                  pbc=np.ones(3, dtype=np.int64))
    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 3.0)
    assert len(g["within_bond_r"]) == 0          # the scenario premise
    core = CHGNetCore.seeded(seed=0).double()
    eng = SpmdEngine(core, world=1, threads=2, device="cpu", ops=CpuRefOps())
    out = eng.step(s)
    ref = oracle_forward(core, s, g["src"], g["dst"], g["offsets"],
                         g["within_bond_r"], dtype=torch.float64)
    assert abs(out["energy"].item() - ref["energy"].item()) < 1e-9
    F = np.zeros((s.num_atoms, 3))
    F[out["global_ids_owned"]] = out["forces_owned"].numpy()
    assert np.abs(F - ref["forces"].numpy()).max() < 1e-10


def test_oracle_rotation_covariance():
    """Oracle physics sanity: rotating the cell leaves the energy
    invariant and rotates forces covariantly (F' = F @ R).  Strengthens
    the oracle pinning: a wrong geometry chain (bond_vec/theta) would
    break this."""
    import torch
    from scipy.stats import ortho_group

    from distmlip_amd.model import CHGNetCore
    from distmlip_amd.structures import random_cell
    from oracle.chgnet_ref import oracle_forward
    from oracle.graph_ref import brute_force_neighbors

    s = random_cell(40, a=11.0, n_species=2, seed=3, skew=0.05)
    core = CHGNetCore.seeded(seed=0)
    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 3.0)
    ref = oracle_forward(core, s, g["src"], g["dst"], g["offsets"],
                         g["within_bond_r"], dtype=torch.float64)

    R = ortho_group.rvs(3, random_state=11)
    s2 = type(s)(frac_coords=s.frac_coords, lattice=s.lattice @ R,
                 species=s.species, pbc=s.pbc)
    g2 = brute_force_neighbors(s2.frac_coords, s2.lattice, s2.pbc, 6.0, 3.0)
    ref2 = oracle_forward(core, s2, g2["src"], g2["dst"], g2["offsets"],
                          g2["within_bond_r"], dtype=torch.float64)

    assert abs(ref["energy"].item() - ref2["energy"].item()) < 1e-9
    FR = ref["forces"].numpy() @ R
    assert np.abs(FR - ref2["forces"].numpy()).max() < 1e-9
