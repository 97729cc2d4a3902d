"""Side-index degradation soak: the store must read IDENTICALLY (for
p>=1 surfaces) under every WK_FN/WK_CSR/WK_TBM/WK_VERSATILE combo and
under tight byte budgets — the side indexes are accelerators, never
semantics.  (p=0 vp lists exist only with WK_VERSATILE=1, mirroring
the reference's USE_VERSATILE build flag.)  Usage: python
tools/soak/env_combo_soak.py"""
import os
import subprocess
import sys

COMBOS = [
    {}, {"WK_FN": "0"}, {"WK_CSR": "0"}, {"WK_TBM": "0"},
    {"WK_VERSATILE": "0"},
    {"WK_FN": "0", "WK_CSR": "0", "WK_TBM": "0", "WK_VERSATILE": "0"},
    {"WK_FN_BUDGET_GB": "1", "WK_CSR_BUDGET_GB": "1", "WK_TBM_BUDGET_GB": "1"},
]

CODE = r'''
import sys, numpy as np
sys.path.insert(0, %r)
import wukong_amd as wk
tri = wk.lubm_gen(2, seed=42)
st = wk.Store(tri)
assert st.check() == 0
rng = np.random.default_rng(3)
subs = np.unique(tri[:, 0])
h = 0
for v in rng.choice(subs, 200, replace=False):
    for p in range(1, 9):
        for d in (0, 1):
            e = st.get_triples(int(v), p, d)
            h = (h * 1000003 + int(e.sum()) + len(e)) %% (1 << 61)
for p in range(1, 9):
    for d in (0, 1):
        e = st.get_index(p, d)
        h = (h * 1000003 + int(np.asarray(e).sum()) + len(e)) %% (1 << 61)
print(h)
'''


def main():
    repo = os.path.dirname(os.path.dirname(
        os.path.dirname(os.path.abspath(__file__))))
    code = CODE % repo
    hashes = []
    for env in COMBOS:
        e = dict(os.environ)
        e.update(env)
        out = subprocess.run([sys.executable, "-c", code],
                             capture_output=True, text=True, env=e)
        if out.returncode != 0:
            print("COMBO FAILED", env, out.stderr[-300:])
            hashes.append(None)
        else:
            hashes.append(out.stdout.strip().splitlines()[-1])
    ok = all(h is not None and h == hashes[0] for h in hashes)
    print("env-combo soak:", "bad=0" if ok else f"DIVERGED {hashes}")
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
