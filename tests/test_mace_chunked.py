"""Node-range-chunked MACE message pass (DM_MACE_CHUNK) must equal the
unchunked pass exactly: same ops over a partition of the dst-sorted
edge ranges, fp64 on CPU."""
import os

import numpy as np
import pytest
import torch


def test_chunked_message_pass_matches_unchunked():
    from distmlip_amd.mace_model import MACEConfig, MACECore
    from distmlip_amd.mace_runtime import MaceSpmdEngine
    from distmlip_amd.structures import diamond_si
    from oracle.chgnet_ref import CpuRefOps

    s = diamond_si((6, 2, 2), jitter=0.1, seed=3)
    s.species = np.asarray(s.species) % 3
    core = MACECore.seeded(MACEConfig(n_elements=3, channels=32),
                           seed=1).double()
    outs = {}
    for tag, chunk in (("un", None), ("ch", "500")):
        if chunk is None:
            os.environ.pop("DM_MACE_CHUNK", None)
        else:
            os.environ["DM_MACE_CHUNK"] = chunk
        try:
            eng = MaceSpmdEngine(core, world=1, threads=4, device="cpu",
                                 ops=CpuRefOps())
            outs[tag] = eng.step(s, calc_stresses=True)
        finally:
            os.environ.pop("DM_MACE_CHUNK", None)
    assert abs(outs["un"]["energy"].item()
               - outs["ch"]["energy"].item()) < 1e-10
    assert (outs["un"]["forces_owned"]
            - outs["ch"]["forces_owned"]).abs().max().item() < 1e-10
    assert (outs["un"]["stress"]
            - outs["ch"]["stress"]).abs().max().item() < 1e-10
