"""Golden-vector KATs — pin the restatements against committed fixtures so
drift is caught without /root/reference (the fixtures were generated in the
build container against the reference's own compiled C module; see
tests/golden/make_golden.py)."""
import json
import os

import numpy as np
import torch

from distmlip_amd.model import CHGNetCore
from distmlip_amd.structures import diamond_si
from oracle.chgnet_ref import build_full_line_graph, oracle_forward
from oracle.graph_ref import brute_force_neighbors

GOLDEN = os.path.join(os.path.dirname(__file__), "golden", "golden.json")


def _load():
    with open(GOLDEN) as f:
        return json.load(f)


def test_golden_graph_shape():
    g0 = _load()
    st = g0["structure"]
    s = diamond_si(tuple(st["reps"]), jitter=st["jitter"], seed=st["seed"])
    assert s.num_atoms == st["n_atoms"]
    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 3.0)
    assert len(g["src"]) == g0["graph"]["n_edges"]
    assert len(g["within_bond_r"]) == g0["graph"]["n_bonds"]
    l_src, l_dst, center = build_full_line_graph(g["src"], g["dst"],
                                                 g["within_bond_r"])
    assert len(l_src) == g0["graph"]["n_lines"]
    assert int(center.sum()) == g0["graph"]["line_center_sum"]
    np.testing.assert_allclose(g["dist"].sum(), g0["graph"]["dist_sum"],
                               rtol=1e-12)


def test_golden_oracle_energy_forces():
    g0 = _load()
    st = g0["structure"]
    s = diamond_si(tuple(st["reps"]), jitter=st["jitter"], seed=st["seed"])
    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 3.0)
    core = CHGNetCore.seeded(seed=g0["model"]["seed"]).double()
    out = oracle_forward(core, s, g["src"], g["dst"], g["offsets"],
                         g["within_bond_r"], dtype=torch.float64)
    np.testing.assert_allclose(out["energy"].item(), g0["model"]["energy"],
                               rtol=1e-10)
    F = out["forces"].numpy()
    np.testing.assert_allclose(np.abs(F).sum(), g0["model"]["forces_abs_sum"],
                               rtol=1e-9)
    np.testing.assert_allclose(F[:3], np.array(g0["model"]["forces_first3"]),
                               rtol=1e-8, atol=1e-10)
    np.testing.assert_allclose(out["site_props"].sum().item(),
                               g0["model"]["site_props_sum"], rtol=1e-9)


def test_golden_reference_partition_shapes(ref_graph_backend):
    """Only runs where the reference oracle is built (build container)."""
    g0 = _load()
    if "reference_partition_P2" not in g0:
        return
    st = g0["structure"]
    s = diamond_si(tuple(st["reps"]), jitter=st["jitter"], seed=st["seed"])
    ref = ref_graph_backend(s.cart_coords, 6.0, s.pbc, s.lattice, 2, 3.0,
                            1e-8, 4, True, s.frac_coords)
    exp = g0["reference_partition_P2"]
    assert [np.asarray(m).tolist() for m in ref[2]] == exp["markers"]
    assert [np.asarray(m).tolist() for m in ref[12]] == exp["line_markers"]
    assert [int(len(np.asarray(e))) for e in ref[0]] == exp["edges_per_partition"]
    assert [int(len(np.asarray(e))) for e in ref[9]] == exp["lines_per_partition"]


def test_golden_mace_uma():
    """Round-2 model families pinned by committed fixtures
    (tests/golden/make_golden_models.py)."""
    from distmlip_amd.mace_model import MACEConfig, MACECore
    from distmlip_amd.uma_model import UMAConfig, UMACore
    from oracle.mace_ref import mace_oracle_forward
    from oracle.uma_ref import uma_oracle_forward

    with open(os.path.join(os.path.dirname(__file__), "golden",
                           "golden_models.json")) as f:
        g0 = json.load(f)
    s = diamond_si((6, 2, 2), jitter=0.1, seed=2)
    s.species = np.asarray(s.species) % 3
    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 0.0)

    mc = MACECore.seeded(MACEConfig(n_elements=3, channels=16,
                                    avg_num_neighbors=20.0,
                                    atomic_inter_scale=0.7,
                                    atomic_inter_shift=0.1),
                         seed=9).double()
    rm = mace_oracle_forward(mc, s, g["src"], g["dst"], g["offsets"])
    assert abs(rm["energy"].item() - g0["mace"]["energy"]) < 1e-9
    np.testing.assert_allclose(rm["forces"][:5].numpy(),
                               np.asarray(g0["mace"]["forces_head"]),
                               rtol=1e-9, atol=1e-10)

    uc = UMACore.seeded(UMAConfig(n_elements=3, sphere_channels=16,
                                  num_layers=2, edge_ch=32, num_gauss=16,
                                  spec_emb=8, avg_degree=20.0),
                        seed=9).double()
    ru = uma_oracle_forward(uc, s, g["src"], g["dst"], g["offsets"])
    assert abs(ru["energy"].item() - g0["uma"]["energy"]) < 1e-9
    np.testing.assert_allclose(ru["forces"][:5].numpy(),
                               np.asarray(g0["uma"]["forces_head"]),
                               rtol=1e-9, atol=1e-10)
