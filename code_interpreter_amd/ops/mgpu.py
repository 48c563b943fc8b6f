"""Multi-GPU helpers for sandboxed user code: torch.distributed over RCCL
(xGMI) inside one executor.

Importable standalone inside the sandbox (APP_OPS_DIR is on sys.path), so
a custom tool can run the BASELINE acceptance workload directly:

    import mgpu
    result = mgpu.allreduce_matmul_bench(size=8192, dtype="bfloat16")

Design notes for the MI355X node:
- one process per GPU, backend "nccl" (= RCCL on ROCm), rendezvous on
  127.0.0.1 with a free port;
- xGMI is point-to-point (7 links x ~153 GB/s per GPU): ring all-reduce is
  per-link bound, so all-reduce buckets default to 64 MB -- large enough
  to amortize launch/latency, small enough to pipeline across links --
  and RCCL's own ring/tree auto-selection does the topology work;
- HSA_ENABLE_IPC_MODE_LEGACY=0 must stay set (dmabuf IPC; inherited from
  the environment).
"""

import os
import socket
from typing import Callable, List, Optional

DEFAULT_BUCKET_BYTES = 64 << 20  # xGMI per-link pipelining sweet spot


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def visible_gpu_count() -> int:
    import torch

    return torch.cuda.device_count() if torch.cuda.is_available() else 0


def _entry(rank: int, world_size: int, port: int, backend: str, fn, args, queue):
    import torch
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    dist.init_process_group(backend=backend, rank=rank, world_size=world_size)
    try:
        if backend == "nccl":
            torch.cuda.set_device(rank % max(1, torch.cuda.device_count()))
            # route this rank's matmuls to the hand-written MFMA kernels
            # (same policy env as the sandbox runtime: APP_HIP_TORCH)
            mode = os.environ.get("APP_HIP_TORCH", "auto").lower()
            if mode != "off":
                try:
                    import sys

                    ops_dir = os.path.dirname(os.path.abspath(__file__))
                    if ops_dir not in sys.path:
                        sys.path.insert(0, ops_dir)
                    import hiptorch

                    hiptorch.install(mode=mode)
                except ImportError:
                    if mode == "require":
                        raise
        result = fn(rank, world_size, *args)
        if rank == 0 and queue is not None:
            queue.put(result)
    finally:
        dist.destroy_process_group()


def run_distributed(
    fn: Callable,
    world_size: Optional[int] = None,
    backend: Optional[str] = None,
    args: tuple = (),
):
    """Run ``fn(rank, world_size, *args)`` on one process per GPU (RCCL
    over xGMI); returns rank 0's return value. With no GPUs (or
    backend="gloo") runs CPU processes -- the same code path the
    multi-process unit tests exercise."""
    import torch.multiprocessing as mp

    if world_size is None:
        world_size = max(1, visible_gpu_count())
    if backend is None:
        backend = "nccl" if visible_gpu_count() > 0 else "gloo"
    port = _free_port()

    ctx = mp.get_context("spawn")
    queue = ctx.SimpleQueue()
    mp.start_processes(
        _entry,
        args=(world_size, port, backend, fn, args, queue),
        nprocs=world_size,
        join=True,
        start_method="spawn",
    )
    return queue.get() if not queue.empty() else None


def allreduce_bucketed(tensors: List, bucket_bytes: Optional[int] = None):
    """All-reduce a list of tensors in flattened buckets sized for xGMI
    ring pipelining (each bucket = one RCCL call; per-link bandwidth bound
    means fewer, larger collectives beat many small ones up to the bucket
    size that still overlaps). bucket_bytes=None picks the size from the
    node topology model (parallel/topology.py): 64 MB default, shrunk so
    small payloads still pipeline across >= 4 in-flight buckets."""
    import torch
    import torch.distributed as dist

    if bucket_bytes is None:
        total = sum(t.numel() * t.element_size() for t in tensors)
        try:
            from code_interpreter_amd.parallel.topology import (
                recommended_bucket_bytes,
            )

            bucket_bytes = recommended_bucket_bytes(
                total, dist.get_world_size()
            )
        except ImportError:  # standalone sandbox import: ops/ only
            bucket_bytes = DEFAULT_BUCKET_BYTES

    bucket: List = []
    used = 0

    def flush():
        nonlocal bucket, used
        if not bucket:
            return
        flat = torch.cat([t.reshape(-1) for t in bucket])
        dist.all_reduce(flat)
        offset = 0
        for t in bucket:
            n = t.numel()
            t.copy_(flat[offset : offset + n].reshape(t.shape))
            offset += n
        bucket, used = [], 0

    for t in tensors:
        nbytes = t.numel() * t.element_size()
        if used and used + nbytes > bucket_bytes:
            flush()
        bucket.append(t)
        used += nbytes
    flush()
    return tensors


def _matmul_allreduce_worker(rank, world_size, size, dtype_name, iters):
    import time

    import torch
    import torch.distributed as dist

    dtype = getattr(torch, dtype_name)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    a = torch.randn(size, size, dtype=dtype, device=device)
    b = torch.randn(size, size, dtype=dtype, device=device)

    # warmup
    c = a @ b
    dist.all_reduce(c)
    if device == "cuda":
        torch.cuda.synchronize()
    dist.barrier()

    t0 = time.perf_counter()
    for _ in range(iters):
        c = a @ b
        dist.all_reduce(c)
    if device == "cuda":
        torch.cuda.synchronize()
    dist.barrier()
    elapsed = (time.perf_counter() - t0) / iters

    matmul_tflops = 2 * size**3 / elapsed / 1e12
    # ring all-reduce moves 2*(N-1)/N of the tensor per GPU
    payload = c.numel() * c.element_size()
    bus_gbps = (
        2 * (world_size - 1) / world_size * payload / elapsed / 1e9
        if world_size > 1
        else 0.0
    )
    return {
        "size": size,
        "dtype": dtype_name,
        "world_size": world_size,
        "device": device,
        "ms_per_iter": round(elapsed * 1e3, 3),
        "matmul_tflops_per_gpu": round(matmul_tflops, 1),
        "allreduce_bus_gbps": round(bus_gbps, 1),
    }


def allreduce_matmul_bench(
    size: int = 8192,
    dtype: str = "bfloat16",
    world_size: Optional[int] = None,
    iters: int = 10,
    backend: Optional[str] = None,
) -> dict:
    """The BASELINE acceptance workload: per-GPU bf16 size^2 matmul with the
    result all-reduced across all GPUs over xGMI. Returns rank 0's stats."""
    return run_distributed(
        _matmul_allreduce_worker,
        world_size=world_size,
        backend=backend,
        args=(size, dtype, iters),
    )
