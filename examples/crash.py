# A payload whose whole point is to raise: the service must return the
# traceback on stderr and a nonzero exit code (never a 500).
values = [1, 2, 3]
print(values[10])
