"""Oracle package — TEST INFRASTRUCTURE ONLY.

This package holds the CPU restatements of the reference's algorithms
(AegisIK/DistMLIP, mounted read-only at /root/reference in the build
container) plus a loader for the reference's own compiled C extension
(`oracle/_ref`, built by `oracle/Makefile` from the reference sources in
place).

Only `tests/`, `__graft_entry__.smoke()` (as the checker) and `bench.py`'s
`cpu_baseline` leg may import, call, link or execute anything under this
package — and there only as the checker / reported CPU baseline, never as
the thing measured or shipped.  The product package (`distmlip_amd/`) must
never import `oracle` and must fail loudly when its HIP extension is
missing.

Parity pinning status:
  * graph layer (neighbor list + partitioner): PINNED — checked against the
    reference's own compiled C module (`oracle/_ref`) and an independent
    O(N^2) brute-force PBC search (`oracle/graph_ref.py`).
  * CHGNet model arithmetic: PARITY UNPINNED at the matgl boundary — the
    reference delegates all model math to matgl (pinned @5171392,
    /root/reference/pyproject.toml:27), which is not installed in this
    container and ships no numeric tests.  `oracle/chgnet_ref.py` is our
    restatement following the reference's own orchestration
    (implementations/matgl/models/chgnet.py:21-453) and the published
    CHGNet architecture; partition-vs-single-graph equality and the
    committed golden vectors are the executable known-answer tests.
"""
