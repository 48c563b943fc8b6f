# First half of the cross-execution file round-trip: writes example.txt,
# which the service detects (ctime scan) and stores; pass the returned
# hash as an input file to hello_world_read_file.py.
from pathlib import Path

Path("example.txt").write_text("hello from a previous execution\n")
