# Write a file into the workspace and read it back in the same execution;
# the service also returns it in `files` (changed-file scan).
from pathlib import Path

Path("note.txt").write_text("written inside the sandbox\n")
print(Path("note.txt").read_text(), end="")
