# Show the sandbox's working directory: a fresh per-execution workspace.
import os

print(os.getcwd())
print(sorted(os.listdir()))
