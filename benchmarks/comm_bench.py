"""Communication / memory micro-benchmarks (reference
benchmarks/tcp_communicator_benchmark.cpp, roce_communicator_benchmark.cpp,
copy_benchmark.cpp, grad_clear_benchmark.cpp analogs, MI355X-native).

Single process:      python benchmarks/comm_bench.py            (copy/clear)
Multi-rank P2P/coll: python -m torch.distributed.run --nnodes=1
                       --nproc-per-node 2 --master-addr 127.0.0.1
                       benchmarks/comm_bench.py

On GPU ranks the P2P plane is RCCL over xGMI (7 links x ~153 GB/s per
GPU); on CPU it is gloo (functional check only). Sizes sweep the
pipeline's real activation messages (WRN-16-8 boundary at batch 256 is
~33.5 MB — SURVEY §2.9 message shapes).
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

SIZES_MB = [1, 4, 16, 33.5, 64, 256]


def _sync(dev):
    if dev.type == "cuda":
        torch.cuda.synchronize()


def bench_copy(dev):
    """H2D / D2H / D2D copy bandwidth (reference copy_benchmark.cpp)."""
    print("== copy bandwidth ==")
    for mb in [16, 64, 256]:
        n = int(mb * 1e6 / 4)
        host = torch.empty(n, pin_memory=(dev.type == "cuda"))
        devt = torch.empty(n, device=dev)
        for name, fn in [("h2d", lambda: devt.copy_(host, non_blocking=True)),
                         ("d2h", lambda: host.copy_(devt, non_blocking=True)),
                         ("d2d", lambda: devt.clone())]:
            fn(); _sync(dev)
            t0 = time.perf_counter()
            for _ in range(10):
                fn()
            _sync(dev)
            dt = time.perf_counter() - t0
            print(f"  {name} {mb:6.1f} MB: {mb * 10 / dt / 1e3:8.1f} GB/s")


def bench_grad_clear(dev):
    """zero_() over a param-sized slab (reference grad_clear_benchmark)."""
    print("== grad clear ==")
    for mb in [17, 124]:  # WRN-16-8 / GPT-2-small param sizes
        t = torch.empty(int(mb * 1e6 / 4), device=dev)
        t.zero_(); _sync(dev)
        t0 = time.perf_counter()
        for _ in range(20):
            t.zero_()
        _sync(dev)
        dt = time.perf_counter() - t0
        print(f"  {mb:4d} MB x20: {mb * 20 / dt / 1e3:8.1f} GB/s")


def bench_p2p(dev):
    import torch.distributed as dist
    rank, world = dist.get_rank(), dist.get_world_size()
    print(f"== P2P ping-pong (rank {rank}/{world}, {dist.get_backend()}) ==")
    for mb in SIZES_MB:
        n = int(mb * 1e6 / 2)
        t = torch.zeros(n, dtype=torch.bfloat16, device=dev)
        reps = 20
        # neighbor pairs (0<->1, 2<->3, ...): the pipeline's traffic shape
        peer = rank + 1 if rank % 2 == 0 else rank - 1
        if peer >= world:
            continue
        dist.barrier()
        _sync(dev)
        t0 = time.perf_counter()
        for _ in range(reps):
            if rank % 2 == 0:
                dist.send(t, peer)
                dist.recv(t, peer)
            else:
                dist.recv(t, peer)
                dist.send(t, peer)
        _sync(dev)
        dt = time.perf_counter() - t0
        bw = 2 * mb * reps / dt / 1e3  # GB/s per direction-pair
        if rank == 0:
            print(f"  {mb:6.1f} MB: {dt / reps * 1e3:7.2f} ms RTT, "
                  f"{bw:8.1f} GB/s")
    # batched bidirectional pair (the 1F1B steady-state op)
    if world >= 2:
        for mb in [33.5]:
            n = int(mb * 1e6 / 2)
            a = torch.zeros(n, dtype=torch.bfloat16, device=dev)
            b = torch.empty_like(a)
            peer = rank + 1 if rank % 2 == 0 else rank - 1
            if peer >= world:
                continue
            dist.barrier()
            _sync(dev)
            t0 = time.perf_counter()
            for _ in range(20):
                reqs = dist.batch_isend_irecv([
                    dist.P2POp(dist.isend, a, peer),
                    dist.P2POp(dist.irecv, b, peer)])
                for r in reqs:
                    r.wait()
            _sync(dev)
            dt = time.perf_counter() - t0
            if rank == 0:
                print(f"  batched bidir {mb:.1f} MB x2: {dt / 20 * 1e3:7.2f} ms"
                      f", {2 * mb * 20 / dt / 1e3:8.1f} GB/s aggregate")


def bench_allreduce(dev):
    import torch.distributed as dist
    rank, world = dist.get_rank(), dist.get_world_size()
    if rank == 0:
        print(f"== all-reduce ({world} ranks) ==")
    for mb in [4, 16, 64, 124]:
        t = torch.zeros(int(mb * 1e6 / 4), device=dev)
        dist.barrier()
        _sync(dev)
        t0 = time.perf_counter()
        for _ in range(10):
            dist.all_reduce(t)
        _sync(dev)
        dt = time.perf_counter() - t0
        # ring algorithm bus bandwidth: 2(n-1)/n * bytes / time
        bus = 2 * (world - 1) / world * mb * 10 / dt / 1e3
        if rank == 0:
            print(f"  {mb:4d} MB: {dt / 10 * 1e3:7.2f} ms, "
                  f"bus {bus:8.1f} GB/s")


def main():
    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    if "RANK" in os.environ:
        import torch.distributed as dist
        backend = "nccl" if dev.type == "cuda" else "gloo"
        dist.init_process_group(backend)
        if dev.type == "cuda":
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
            dev = torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0)))
        bench_p2p(dev)
        bench_allreduce(dev)
        dist.destroy_process_group()
    if os.environ.get("RANK", "0") == "0":
        bench_copy(dev)
        bench_grad_clear(dev)


if __name__ == "__main__":
    main()
