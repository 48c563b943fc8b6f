# CPU-bound workload (shape-compatible with the reference's
# examples/benchmark-fib.py): 1000 iterative fib(10000) runs.
import time

def compute():
    def fib(n):
        a, b = 0, 1
        for _ in range(n):
            a, b = b, a + b
        return a
    return sum(fib(10000) for _ in range(1000))

start = time.time()
compute()
print("Execution Time:", time.time() - start, "seconds")
