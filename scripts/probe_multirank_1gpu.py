"""Probe: do 2 RCCL ranks on ONE MI355X work?

Launched as:  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2
              --master-addr 127.0.0.1 --master-port 29571
              scripts/probe_multirank_1gpu.py

Both ranks pin cuda:0.  Exercises, in order: init_process_group(nccl),
broadcast, all_gather_into_tensor, all_to_all_single, all_reduce — the
exact collectives DistributedRuntime uses — then a full 2-rank Simulator
round (full gather AND coordinate gather) and prints per-rank OK lines.
"""
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    rank = int(os.environ["RANK"])
    torch.cuda.set_device(0)  # both ranks share the single GPU
    os.environ["LOCAL_RANK"] = "0"
    dist.init_process_group("nccl")
    dev = torch.device("cuda:0")

    # raw collectives
    t = torch.full((1024,), float(rank + 1), device=dev)
    dist.broadcast(t, src=0)
    assert t.mean().item() == 1.0, t.mean().item()

    out = torch.empty(2 * 1024, device=dev)
    dist.all_gather_into_tensor(out, torch.full((1024,), float(rank), device=dev))
    assert out[:1024].mean().item() == 0.0 and out[1024:].mean().item() == 1.0

    s = torch.arange(8, dtype=torch.float32, device=dev) + 10 * rank
    r = torch.empty_like(s)
    dist.all_to_all_single(r, s)
    # rank r receives [r*4..r*4+4) from rank 0 and same from rank 1 (+10)
    exp = torch.cat([torch.arange(rank * 4, rank * 4 + 4),
                     torch.arange(rank * 4, rank * 4 + 4) + 10]).float().to(dev)
    assert torch.equal(r, exp), (r, exp)

    a = torch.ones(64, device=dev)
    dist.all_reduce(a)
    assert a[0].item() == 2.0
    print(f"[rank {rank}] raw RCCL collectives OK", flush=True)

    # full simulator round, both gathers
    from blades_amd import Simulator
    from blades_amd.datasets import SyntheticFLDataset
    from blades_amd.models import MLP

    for gather in ("full", "coordinate"):
        ds = SyntheticFLDataset(num_clients=8, samples_per_client=16,
                                batch_size=8, shape=(1, 28, 28),
                                num_classes=10, seed=0, device="cuda:0")
        sim = Simulator(ds, num_byzantine=2, attack="alie",
                        attack_kws={"num_clients": 8, "num_byzantine": 2},
                        aggregator="trimmedmean", aggregator_kws={"nb": 2},
                        use_cuda=True, device="cuda:0",
                        log_path=f"/tmp/probe_mr_{gather}_{rank}",
                        seed=0, gather=gather)
        ret = sim.run(MLP(), global_rounds=3, local_steps=1, client_lr=0.1,
                      server_lr=1.0, validate_interval=0)
        theta = sim.server.flat_parameters()
        assert torch.isfinite(theta).all()
        print(f"[rank {rank}] 2-rank round gather={gather} OK "
              f"theta[:3]={theta[:3].tolist()}", flush=True)

    dist.barrier()
    dist.destroy_process_group()
    print(f"[rank {rank}] ALL OK", flush=True)


if __name__ == "__main__":
    main()
