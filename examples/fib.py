# Naive exponential-recursion fibonacci: pure-CPU sandbox workload with
# no imports at all (shape-compatible with the reference's examples/fib.py).
def fib(n: int) -> int:
    return n if n < 2 else fib(n - 1) + fib(n - 2)

for i in range(30):
    print(i, fib(i))
