"""ASE integration mirror — PESCalculator_Dist / Relaxer / MolecularDynamics.

API mirror of the reference implementations/matgl/ase.py:53-490 (thin
wrappers over ASE drivers).  ase/pymatgen are OPTIONAL (not installed in
the build container; reference pins them as extras, pyproject.toml:26-35):
imports are lazy and fail with a clear message.  Units follow ASE (eV,
eV/A; stress scale handled in pes.py, -160.21766208 eV/A^3, pes.py:143).
"""
from __future__ import annotations

import numpy as np

from distmlip_amd.pes import Potential_Dist
from distmlip_amd.structures import Structure


def _require_ase():
    try:
        import ase  # noqa: F401
    except ImportError as e:
        raise ImportError(
            "ase is required for the calculator/MD layer (optional "
            "dependency, reference pyproject.toml:26-35)") from e


class PESCalculator_Dist:
    """Mirror of reference ase.py:53-127 (ASE Calculator over Potential_Dist).

    Implemented as a factory returning a genuine ase Calculator subclass
    (created lazily so the module imports without ase)."""

    def __new__(cls, potential: Potential_Dist, state_attr=None, **kwargs):
        _require_ase()
        from ase.calculators.calculator import Calculator, all_changes

        pot = potential

        class _Calc(Calculator):
            implemented_properties = ["energy", "free_energy", "forces",
                                      "stress", "magmoms"]

            def __init__(self):
                super().__init__(**kwargs)
                self.potential = pot
                self.element_to_index = {
                    str(el): i for i, el in enumerate(
                        range(pot.model.core.config.n_elements))}

            def calculate(self, atoms=None, properties=None,
                          system_changes=all_changes):
                properties = properties or ["energy"]
                super().calculate(atoms, properties, system_changes)
                # species mapping: caller supplies element_to_index via the
                # model; default = atomic numbers - 1
                e2i = getattr(pot.model, "element_to_index", None)
                if e2i is not None:
                    species = np.array([e2i[s] for s in
                                        atoms.get_chemical_symbols()])
                    st = Structure(
                        frac_coords=atoms.get_scaled_positions(wrap=True),
                        lattice=np.array(atoms.get_cell()),
                        species=species.astype(np.int64),
                        pbc=atoms.get_pbc().astype(np.int64))
                else:
                    st = Structure.from_ase(
                        atoms, {s: z - 1 for s, z in zip(
                            atoms.get_chemical_symbols(),
                            atoms.get_atomic_numbers())})
                E, F, S, _ = self.potential.forward(st)
                self.results["energy"] = float(E.detach().cpu())
                self.results["free_energy"] = self.results["energy"]
                if F is not None:
                    self.results["forces"] = F.detach().cpu().numpy()
                if S is not None:
                    from ase.stress import full_3x3_to_voigt_6_stress
                    self.results["stress"] = full_3x3_to_voigt_6_stress(
                        S.detach().cpu().numpy())

        return _Calc()


class Relaxer:
    """Mirror of reference ase.py:130-223 (structure relaxation driver)."""

    def __init__(self, potential: Potential_Dist, optimizer: str = "FIRE",
                 relax_cell: bool = True):
        _require_ase()
        self.calculator = PESCalculator_Dist(potential)
        self.optimizer_name = optimizer
        self.relax_cell = relax_cell

    def relax(self, atoms, fmax: float = 0.1, steps: int = 500,
              traj_file=None, interval: int = 1, **kwargs):
        import ase.optimize
        from ase.constraints import ExpCellFilter

        atoms.calc = self.calculator
        target = ExpCellFilter(atoms) if self.relax_cell else atoms
        opt_cls = getattr(ase.optimize, self.optimizer_name)
        opt = opt_cls(target, trajectory=traj_file, **kwargs)
        opt.run(fmax=fmax, steps=steps)
        return {"final_structure": atoms,
                "energy": self.calculator.results.get("energy")}


class MolecularDynamics:
    """Mirror of reference ase.py:228-490 (NVE/NVT/NPT MD driver)."""

    def __init__(self, atoms, potential: Potential_Dist,
                 ensemble: str = "nvt", temperature: float = 300.0,
                 timestep: float = 1.0, pressure: float = None,
                 taut: float = None, logfile=None, loginterval: int = 1,
                 trajectory=None):
        _require_ase()
        import ase.units as units
        from ase.md.nvtberendsen import NVTBerendsen
        from ase.md.npt import NPT
        from ase.md.verlet import VelocityVerlet

        atoms.calc = PESCalculator_Dist(potential)
        self.atoms = atoms
        kw = dict(timestep=timestep * units.fs, logfile=logfile,
                  loginterval=loginterval, trajectory=trajectory)
        ens = ensemble.lower()
        if ens == "nve":
            self.dyn = VelocityVerlet(atoms, **kw)
        elif ens == "nvt":
            self.dyn = NVTBerendsen(
                atoms, temperature_K=temperature,
                taut=taut or (100 * units.fs), **kw)
        elif ens == "npt":
            self.dyn = NPT(atoms, temperature_K=temperature,
                           externalstress=(pressure or 0.0), **kw)
        else:
            raise ValueError(f"unknown ensemble {ensemble!r}")

    def run(self, steps: int):
        self.dyn.run(steps)
