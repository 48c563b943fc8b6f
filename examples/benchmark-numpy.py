# The headline benchmark workload (shape-compatible with the reference's
# examples/benchmark-numpy.py): 1e8-element uniform array -> square -> sum.
# Inside an MI355X sandbox the three numpy calls run on the gfx950 HIP
# kernels (Philox RNG + fused square+sum) without leaving device memory.
import numpy
import time

def compute():
    array_size = 10**8
    large_array = numpy.random.rand(array_size)
    return numpy.sum(numpy.square(large_array))

start_time = time.time()
result = compute()
end_time = time.time()
print("Result:", result)
print("Execution Time:", end_time - start_time, "seconds")
