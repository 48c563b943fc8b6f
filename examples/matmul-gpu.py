# fp32 matmul routed to the MFMA matrix cores (v_mfma_f32_32x32x2_f32).
# Host-born operands: the timing includes staging 128 MB into the GPU
# daemon over the /dev/shm fast path (device-born chains, like
# benchmark-numpy.py, skip staging entirely).
import numpy as np
import time

a = np.random.uniform(-1, 1, (4096, 4096)).astype(np.float32)
b = np.random.uniform(-1, 1, (4096, 4096)).astype(np.float32)
t0 = time.time()
c = np.matmul(a, b)
checksum = float(np.sum(c))
dt = time.time() - t0
print("kind:", type(c).__name__)
print(f"4096^3 fp32 matmul: {dt*1000:.1f} ms = {2*4096**3/dt/1e12:.1f} TFLOP/s")
