"""On-box analysis of nohalo1 broken-state dumps (cpu vs gpu twins)."""
import sys
import numpy as np

base = sys.argv[1]          # e.g. gpurun_out/nh1

def load(trial, dev, name, r):
    return np.load(f"{base}_{trial}/{dev}_{trial}/{name}_{r}.npy")

# --- argmin / vec forward comparison + error-row correlation
for r in range(2):
    vg = load("arg", "gpu", "vec", r)
    vc = load("arg", "cpu", "vec", r)
    ag = load("arg", "gpu", "argmin", r)
    ac = load("arg", "cpu", "argmin", r)
    sd = load("arg", "gpu", "srcdst", r)
    pg = load("arg", "gpu", "prehalo", r)
    pc = load("arg", "cpu", "prehalo", r)
    pe = np.abs(pg - pc).max(axis=1)
    bad_rows = set(np.where(pe > 0.05)[0].tolist())
    flips = np.where(ag != ac)[0]
    print(f"r{r}: vec maxdiff {np.abs(vg-vc).max():.2e}  "
          f"argmin flips {len(flips)}/{len(ag)}  "
          f"prehalo rows>0.05: {len(bad_rows)}")
    if len(flips):
        touch = set(sd[0, flips].tolist()) | set(sd[1, flips].tolist())
        print(f"   flip edges touch {len(touch)} rows; "
              f"overlap with bad rows: {len(touch & bad_rows)}; "
          f"bad rows total {len(bad_rows)}")
        # near-tie margin on flipped edges
        m = np.sort(np.abs(vg[flips]), axis=1)
        print("   flip margins (|v|2nd-|v|1st):",
              np.round((m[:, 1] - m[:, 0])[:8], 8).tolist())

# --- broken-state grad dumps: compare where the WRONG rank's grads differ
for trial, name, r in (("hv", "grad_vectors", 1), ("hD", "grad_D", 0),
                       ("hx", "grad_xedge", 1)):
    try:
        g = load(trial, "gpu", name, r)
        c = load(trial, "cpu", name, r)
    except FileNotFoundError as e:
        print(f"{trial} r{r}: missing {e}")
        continue
    d = np.abs(g - c).reshape(len(g), -1).max(axis=1)
    scale = np.abs(c).max()
    print(f"{trial} {name} r{r}: maxdiff {d.max():.4f} (scale {scale:.3f}) "
          f"edges>1e-3: {(d > 1e-3).sum()}/{len(d)}")
    worst = d.argsort()[-6:][::-1]
    print(f"   worst edges {worst.tolist()} diffs "
          f"{np.round(d[worst], 4).tolist()}")
    sd = load("arg", "gpu", "srcdst", r)
    if len(sd[0]) == len(d):
        print(f"   worst src {sd[0][worst].tolist()} "
              f"dst {sd[1][worst].tolist()}")
