"""RCCL hardware smoke: validate torch.distributed
over the nccl(=RCCL) backend plus the GradReducer bucketed all-reduce on a
real MI355X before the driver's first 8-GPU run.

Run under torchrun; with a single GPU, 2 ranks share device 0 (RCCL may
refuse same-device communicators -- that outcome is reported, not fatal;
world_size=1 still exercises init + broadcast + all-reduce + reducer).

    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --standalone --local-addr 127.0.0.1 scripts/rccl_smoke.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from pvraft_amd.data import synthetic_batch
from pvraft_amd.model import PVRaft
from pvraft_amd.parallel import GradReducer, broadcast_module
from pvraft_amd.utils import sequence_loss


def main():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    n_dev = torch.cuda.device_count()
    dev = torch.device(f"cuda:{rank % n_dev}")
    torch.cuda.set_device(dev)
    try:
        dist.init_process_group("nccl")
    except Exception as e:
        print(f"[rank {rank}] nccl init FAILED: {e!r}")
        return 1
    t = torch.full((1024,), float(rank + 1), device=dev)
    dist.all_reduce(t)
    expect = world * (world + 1) / 2
    assert torch.allclose(t, torch.full_like(t, expect)), t[:4]
    print(f"[rank {rank}] nccl all_reduce OK (world={world}, dev={dev})")

    torch.manual_seed(100 + rank)
    model = PVRaft(truncate_k=64).to(dev)
    broadcast_module(model)
    reducer = GradReducer(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    batch = synthetic_batch(1, 1024, device=dev, seed=rank)
    for _ in range(2):
        reducer.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            flows = model(batch["sequence"], num_iters=2)
            loss = sequence_loss(flows, batch, gamma=0.8)
        loss.backward()
        reducer.finalize()
        opt.step()
    # ranks must agree after synchronous updates
    flat = torch.cat([p.detach().flatten() for p in model.parameters()])
    flats = [torch.empty_like(flat) for _ in range(world)]
    dist.all_gather(flats, flat)
    for f in flats[1:]:
        assert torch.allclose(flats[0], f, atol=1e-5)
    print(f"[rank {rank}] GradReducer train steps over RCCL OK, params in sync")
    dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
