# Second half of the round-trip: expects example.txt to have been staged
# into the workspace via the `files` request field.
from pathlib import Path

print(Path("example.txt").read_text(), end="")
