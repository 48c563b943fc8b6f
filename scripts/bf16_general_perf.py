import sys, time
sys.path.insert(0, "/root/repo/code_interpreter_amd/ops")
import numpy as np
import _hipops
_hipops.init(0)
def bench(m, n, k):
    a = (np.random.uniform(-1,1,(m,k)).astype(np.float32).view(np.uint32) >> 16).astype(np.uint16)
    b = (np.random.uniform(-1,1,(k,n)).astype(np.float32).view(np.uint32) >> 16).astype(np.uint16)
    ha, hb = _hipops.upload(a), _hipops.upload(b)
    for _ in range(3): _hipops.free(_hipops.gemm(ha, hb, m, n, k, 2))
    _hipops.synchronize()
    t0 = time.perf_counter()
    for _ in range(10): _hipops.free(_hipops.gemm(ha, hb, m, n, k, 2))
    _hipops.synchronize()
    dt = (time.perf_counter()-t0)/10
    print(f"bf16 {m}x{n}x{k}: {dt*1e3:.3f} ms {2*m*n*k/dt/1e12:.0f} TF")
    _hipops.free(ha); _hipops.free(hb)
for shape in [(4096,4096,4096),(4000,4000,4000),(4096,4096,4064),(2000,3000,1000),(8000,8000,8000)]:
    bench(*shape)
