"""AD-PSGD device-resident comm (SURVEY C10 MI355X mode): shared CUDA
tensors via dmabuf IPC + RCCL rank owned by the gossip process.

The full 2-rank RCCL consensus needs 2 GPUs (NCCL forbids two ranks on
one device) and auto-skips on a 1-GPU box; the IPC sharing machinery and
the on-device gossip-side fused SGD are validated on 1 GPU.
"""

import time

import pytest
import torch
import torch.multiprocessing as mp
import torch.nn as nn

from tests.dist_utils import free_port

pytestmark = pytest.mark.gpu


def dev():
    return torch.device("cuda", 0)


def _ipc_child(q_in, q_out):
    t = q_in.get()  # dmabuf-IPC view of the parent's device memory
    with torch.no_grad():
        t.add_(1.0)
        torch.cuda.synchronize(t.device)
    q_out.put("done")


def test_cuda_ipc_tensor_sharing():
    """A CUDA tensor sent through an mp.Queue maps the SAME device
    allocation in the child (requires dmabuf IPC; the AD-PSGD
    trainer<->gossip hand-off depends on this)."""
    ctx = mp.get_context("spawn")
    q_in, q_out = ctx.Queue(), ctx.Queue()
    p = ctx.Process(target=_ipc_child, args=(q_in, q_out), daemon=True)
    p.start()
    x = torch.full((1 << 12,), 2.0, device=dev())
    torch.cuda.synchronize()
    q_in.put(x)
    assert q_out.get(timeout=60) == "done"
    p.join(timeout=30)
    torch.cuda.synchronize()
    assert torch.all(x == 3.0), x.unique()


def _gpu_trainer(rank, world_size, port):
    from stochastic_gradient_push_amd import BilatGossipDataParallel
    from stochastic_gradient_push_amd.graphs import (
        DynamicBipartiteExponentialGraph,
    )

    torch.cuda.set_device(rank)

    def tiny(seed):
        torch.manual_seed(seed)
        return nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))

    model = tiny(seed=rank).cuda(rank)
    flats = []
    for s in range(world_size):
        m = tiny(seed=s)
        flats.append(torch.cat([
            p.detach().reshape(-1) for p in m.parameters()
        ]))
    target = torch.stack(flats).mean(0).cuda(rank)

    bgdp = BilatGossipDataParallel(
        model,
        master_addr="127.0.0.1",
        master_port=port,
        backend="nccl",
        world_size=world_size,
        rank=rank,
        graph_class=DynamicBipartiteExponentialGraph,
        comm_device=torch.device("cuda", rank),
        lr=0.0, momentum=0.0, weight_decay=0.0, nesterov=False,
    )
    bgdp.train()
    x = torch.randn(2, 8, device=rank)
    y = torch.randint(0, 4, (2,), device=rank)
    loss_fn = nn.CrossEntropyLoss()
    deadline = time.time() + 120
    converged = False
    while time.time() < deadline:
        loss = loss_fn(bgdp(x), y)
        loss.backward()
        bgdp.sync_comms()
        flat = bgdp.flatp.flat.detach()
        assert torch.isfinite(flat).all()
        if torch.allclose(flat, target, atol=1e-3):
            converged = True
            break
        time.sleep(0.05)
    assert converged, (
        f"rank {rank}: max err {(bgdp.flatp.flat - target).abs().max()}"
    )


@pytest.mark.skipif(
    not torch.cuda.is_available() or torch.cuda.device_count() < 2,
    reason="device-comm consensus needs 2 GPUs (one RCCL rank per GPU)",
)
def test_adpsgd_cuda_consensus():
    """comm_device=cuda end to end: gossip processes own RCCL ranks and
    bilaterally average device-resident parameters."""
    port = free_port()
    ctx = mp.get_context("spawn")
    procs = [
        ctx.Process(target=_gpu_trainer, args=(r, 2, port))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
