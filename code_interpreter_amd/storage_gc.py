"""Object-storage garbage collection.

The workspace file store grows without bound (the reference README
delegates GC to consumers, e.g. S3 TTL rules). For filesystem deployments
this utility deletes objects whose mtime is older than a TTL:

    python -m code_interpreter_amd.storage_gc --ttl-hours 24 [--dry-run]
"""

import argparse
import os
import re
import time

from code_interpreter_amd.config import Config

_OBJECT_RE = re.compile(r"^[0-9a-f]{64}$")


def collect(storage_path: str, ttl_hours: float, dry_run: bool = False) -> dict:
    cutoff = time.time() - ttl_hours * 3600
    removed = kept = freed = 0
    if not os.path.isdir(storage_path):
        return {"removed": 0, "kept": 0, "freed_bytes": 0}
    for name in os.listdir(storage_path):
        if not _OBJECT_RE.match(name):
            continue
        path = os.path.join(storage_path, name)
        try:
            st = os.stat(path)
        except OSError:
            continue
        if st.st_mtime < cutoff:
            if not dry_run:
                try:
                    os.unlink(path)
                except OSError:
                    continue
            removed += 1
            freed += st.st_size
        else:
            kept += 1
    return {"removed": removed, "kept": kept, "freed_bytes": freed}


def main() -> None:
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument("--ttl-hours", type=float, default=24.0)
    parser.add_argument("--storage-path", default=None)
    parser.add_argument("--dry-run", action="store_true")
    args = parser.parse_args()
    path = args.storage_path or Config().file_storage_path
    stats = collect(path, args.ttl_hours, args.dry_run)
    verb = "would remove" if args.dry_run else "removed"
    print(
        f"{verb} {stats['removed']} objects "
        f"({stats['freed_bytes']} bytes), kept {stats['kept']}"
    )


if __name__ == "__main__":
    main()
