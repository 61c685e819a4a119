"""BASELINE config 1: 2-rank tile notify/wait + all-gather correctness on
CPU/gloo — plumbing tests of the symmetric heap, no GPU.

Mirrors the semantics of the reference's canonical primitive test
(Triton-distributed test/amd/test_distributed-notify-wait.py:36-90 — a
cross-rank SPSC queue over symmetric memory) on the CPU mock heap.
"""
import torch

from tests.conftest import run_distributed


def _body_views(rank, world):
    from triton_dist_amd.runtime.symm_mem import SymmHeap

    heap = SymmHeap()
    buf = heap.alloc_buffer((4,), torch.int32)
    buf.local()[:] = rank * 10 + torch.arange(4, dtype=torch.int32)
    heap.barrier_all()
    for r in range(world):
        expect = r * 10 + torch.arange(4, dtype=torch.int32)
        assert torch.equal(buf.peer(r), expect), (rank, r, buf.peer(r))
    heap.barrier_all()
    heap.close()


def test_symm_views_2rank():
    run_distributed(_body_views, world_size=2)


def _body_notify_wait(rank, world):
    """SPSC ring queue: rank 0 produces into rank 1's queue, signals a flag
    per slot; rank 1 waits per slot, checks payload, acks back."""
    from triton_dist_amd.runtime import cpu_shm
    from triton_dist_amd.runtime.symm_mem import SymmHeap

    heap = SymmHeap()
    depth, rounds = 4, 32
    queue = heap.alloc_buffer((depth, 16), torch.float32)
    full = heap.alloc_buffer((depth,), torch.int32)   # producer -> consumer
    empty = heap.alloc_buffer((depth,), torch.int32)  # consumer -> producer
    heap.barrier_all()

    if rank == 0:
        peer_q = queue.peer(1)
        peer_full = full.peer(1)
        my_empty = empty.local()
        for i in range(rounds):
            slot = i % depth
            if i >= depth:  # wait for consumer ack (epoch = #acks for slot)
                cpu_shm.wait_ge(my_empty, slot, i // depth)
            peer_q[slot] = float(i) * torch.ones(16)
            cpu_shm.notify(peer_full, slot, i + 1)
    else:
        my_q = queue.local()
        my_full = full.local()
        peer_empty = empty.peer(0)
        for i in range(rounds):
            slot = i % depth
            cpu_shm.wait_ge(my_full, slot, i + 1)
            assert torch.equal(my_q[slot], float(i) * torch.ones(16)), i
            cpu_shm.notify(peer_empty, slot, i // depth + 1)
    heap.barrier_all()
    heap.close()


def test_notify_wait_ring_2rank():
    run_distributed(_body_notify_wait, world_size=2)


def _body_allgather(rank, world):
    """One-sided push all-gather through the symmetric heap, checked against
    torch.distributed.all_gather (the golden reference, like the reference
    repo's --check mode)."""
    import torch.distributed as dist

    from triton_dist_amd.runtime import cpu_shm
    from triton_dist_amd.runtime.symm_mem import SymmHeap

    heap = SymmHeap()
    m, k = 32, 16
    g = torch.Generator().manual_seed(1234 + rank)
    local = torch.randn(m, k, generator=g)

    ws = heap.alloc_buffer((world, m, k), torch.float32)
    flags = heap.alloc_buffer((world,), torch.int32)
    heap.barrier_all()

    # push my shard into every peer's workspace, then signal
    for r in range(world):
        ws.peer(r)[rank].copy_(local)
        cpu_shm.notify(flags.peer(r), rank, 1)
    # consume: wait for each shard flag
    me = flags.local()
    for r in range(world):
        cpu_shm.wait_ge(me, r, 1)
    gathered = ws.local().clone()

    golden = [torch.empty_like(local) for _ in range(world)]
    dist.all_gather(golden, local)
    assert torch.equal(gathered, torch.stack(golden))
    heap.barrier_all()
    heap.close()


def test_allgather_push_2rank():
    run_distributed(_body_allgather, world_size=2)


def test_allgather_push_4rank():
    run_distributed(_body_allgather, world_size=4)


def _body_barrier_stress(rank, world):
    from triton_dist_amd.runtime.symm_mem import SymmHeap

    heap = SymmHeap()
    counter = heap.alloc_buffer((1,), torch.int32)
    for i in range(50):
        if rank == i % world:
            for r in range(world):
                counter.peer(r)[0] = i
        heap.barrier_all()
        assert int(counter.local()[0]) == i, (rank, i)
        heap.barrier_all()
    heap.close()


def test_barrier_stress_2rank():
    run_distributed(_body_barrier_stress, world_size=2)
