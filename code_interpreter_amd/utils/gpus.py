"""GPU topology detection for executor scheduling (no torch import: the
control plane stays light; detection reads KFD sysfs / rocm-smi)."""

import functools
import glob
import os
import subprocess

KFD_NODES = "/sys/class/kfd/kfd/topology/nodes"


@functools.lru_cache(maxsize=1)
def detect_gpu_count() -> int:
    """Number of visible AMD GPUs. Honors HIP_VISIBLE_DEVICES. Returns 0 in
    CPU-only environments."""
    visible = os.environ.get("HIP_VISIBLE_DEVICES")
    physical = _physical_gpu_count()
    if visible is not None:
        if visible.strip() == "":
            return 0
        ids = [x for x in visible.split(",") if x.strip() != ""]
        return min(len(ids), physical) if physical else 0
    return physical


def _physical_gpu_count() -> int:
    # KFD topology: GPU nodes have non-zero simd_count
    count = 0
    for props in glob.glob(KFD_NODES + "/*/properties"):
        try:
            with open(props) as f:
                for line in f:
                    if line.startswith("simd_count"):
                        if int(line.split()[1]) > 0:
                            count += 1
                        break
        except OSError:
            continue
    if count:
        return count
    # fallback: rocm-smi
    try:
        out = subprocess.run(
            ["rocm-smi", "--showid"],
            capture_output=True,
            text=True,
            timeout=10,
        ).stdout
        return sum(1 for line in out.splitlines() if line.strip().startswith("GPU["))
    except (OSError, subprocess.SubprocessError):
        return 0
