"""A/B validation + perf for the bf16 256^2 kernel variants."""
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

CHECK = r'''
import sys, numpy as np
sys.path.insert(0, "code_interpreter_amd/ops")
import _hipops
_hipops.init(0)

def to_bf16(a):
    u = a.astype(np.float32).view(np.uint32)
    lsb = (u >> 16) & 1
    return ((u + 0x7FFF + lsb) >> 16).astype(np.uint16)

def from_bf16(b):
    return (b.astype(np.uint32) << 16).view(np.float32)

bad = 0
for rep in range(3):
    for size in (256, 512, 768, 4096):
        rng = np.random.default_rng(100 * rep + size)
        a = to_bf16(rng.uniform(-1, 1, (size, size)))
        b = to_bf16(rng.uniform(-1, 1, (size, size)))
        ha, hb = _hipops.upload(a), _hipops.upload(b)
        hc = _hipops.gemm(ha, hb, size, size, size, 2)
        out = np.empty((size, size), dtype=np.uint16)
        _hipops.download(hc, out)
        for h in (ha, hb, hc): _hipops.free(h)
        ref = from_bf16(a).astype(np.float64) @ from_bf16(b).astype(np.float64)
        got = from_bf16(out).astype(np.float64)
        err = np.max(np.abs(got - ref) / (np.abs(ref) + 1.0))
        ok = err < 0.02
        bad += not ok
        print(f"rep{rep} {size}^3 relerr={err:.5f} {'OK' if ok else 'FAIL'}")
sys.exit(1 if bad else 0)
'''

for variant in ("t", "b"):
    env = dict(os.environ, APP_BF16_256_VARIANT=variant)
    r = subprocess.run([sys.executable, "-c", CHECK], env=env, cwd=REPO,
                       capture_output=True, text=True, timeout=600)
    tail = "\n".join(r.stdout.strip().splitlines()[-3:])
    print(f"== variant {variant}: numerics rc={r.returncode}\n{tail}")
    if r.returncode != 0:
        print(r.stdout, r.stderr[-500:])
        continue
    for size in (4096, 8192):
        p = subprocess.run(
            [sys.executable, "scripts/gemm_prof.py", "bf16", str(size)],
            env=env, cwd=REPO, capture_output=True, text=True, timeout=600)
        print(f"   perf {size}: {p.stdout.strip() or p.stderr[-200:]}")
