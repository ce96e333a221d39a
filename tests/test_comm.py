"""Unit tests for the RCCL/gloo communicator (parallel/comm.py): fused
flat-buffer gradient averaging, parameter broadcast, scalar all-reduce —
world_size 2 over gloo (SURVEY §2.2 collective inventory)."""
import os

import numpy as np
import torch


def _worker(rank, world, port):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "LOCAL_RANK": str(rank),
    })
    from deepconsensus_amd.parallel import comm

    r, w = comm.init_distributed(backend="gloo")
    assert (r, w) == (rank, world)

    torch.manual_seed(100 + rank)  # deliberately different init per rank
    net = torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.Linear(8, 2))
    comm.broadcast_parameters(net)
    # After broadcast all ranks hold rank 0's weights.
    torch.manual_seed(100)
    ref = torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.Linear(8, 2))
    for p, q in zip(net.parameters(), ref.parameters()):
        assert torch.equal(p, q)

    reducer = comm.FlatGradAllreducer(net)
    reducer.zero_()
    # p.grad tensors are views into the flat buffer: autograd ACCUMULATES
    # into them (never reassign p.grad — fill in place like backward does).
    for p in net.parameters():
        p.grad.fill_(float(rank + 1))
    reducer.reduce()
    expect = sum(range(1, world + 1)) / world
    for p in net.parameters():
        assert torch.allclose(p.grad, torch.full_like(p, expect)), (
            rank, p.grad.flatten()[0]
        )

    total = comm.allreduce_scalar(float(rank + 1))
    assert abs(total - sum(range(1, world + 1))) < 1e-6
    comm.barrier()
    import torch.distributed as dist

    dist.destroy_process_group()


def test_comm_collectives_gloo_world2():
    torch.multiprocessing.spawn(_worker, args=(2, 29881), nprocs=2,
                                join=True)


def test_flat_reducer_rebinds_detached_grads():
    """A reassigned p.grad is folded back into the bucket on reduce()."""
    from deepconsensus_amd.parallel import comm

    net = torch.nn.Linear(3, 3)
    reducer = comm.FlatGradAllreducer(net)
    reducer.zero_()
    net.weight.grad = torch.full_like(net.weight, 2.0)  # detached
    net.bias.grad = None
    reducer.reduce()  # world 1: no collective, but rebinding runs
    assert net.weight.grad.data_ptr() == reducer._views[0].data_ptr()
    assert torch.all(net.weight.grad == 2.0)
    assert net.bias.grad is not None
    assert float(reducer.grad_norm()) > 0
