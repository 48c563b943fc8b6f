# Multi-GPU custom-tool workload: per-GPU bf16 8192^2 matmul with the
# result all-reduced over xGMI (RCCL), one process per visible GPU.
import mgpu

stats = mgpu.allreduce_matmul_bench(size=8192, dtype="bfloat16")
print(stats)
