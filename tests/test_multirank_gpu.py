"""Hardware tests for the multi-rank path on a 1-GPU box.

RCCL refuses two ranks on one device ("Duplicate GPU detected", probed in
round 2 — gpurun_out/r2_gputest1.log) and CPX partitioning is blocked on
this pool (gpurun_out/cpx_probe.log), so a single-GPU lease cannot run a
true 2-rank RCCL job.  What IS hardware-proven here:

* the RCCL backend executes every collective the runtime uses at ws=1
  INSIDE hipGraph capture+replay (scripts/probe_rccl_graph.py — validates
  RCCL stream-capture support, the mechanism multi-rank CapturedRound
  relies on);
* a REAL 2-process round with all client math on the GPU, collectives
  staged over gloo (BLADES_AMD_BACKEND=gloo) — the full multi-rank round
  logic (shard split, slab layout, all-gather/all-to-all re-shard,
  rank-local attacks, shard aggregation) runs on hardware and must
  reproduce the single-rank θ.
"""
import os
import subprocess
import sys

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
def test_rccl_collectives_inside_hipgraph_ws1(tmp_path):
    # file-backed output (not pipes) + explicit kill so a hang shows its
    # last completed step instead of a bare TimeoutExpired
    log = tmp_path / "probe.log"
    with open(log, "w") as f:
        p = subprocess.Popen(
            [sys.executable, "-u",
             os.path.join(REPO, "scripts", "probe_rccl_graph.py")],
            stdout=f, stderr=subprocess.STDOUT, cwd=REPO,
            stdin=subprocess.DEVNULL)
        try:
            rc = p.wait(timeout=240)
        except subprocess.TimeoutExpired:
            p.kill()
            p.wait(timeout=30)
            rc = None
    out = open(log).read()
    # the assertion is about RCCL stream-capture FUNCTIONALITY: every
    # capture+replay step must have executed and verified.  The child's
    # process EXIT can deadlock under a pytest parent even after
    # os._exit (observed only in-suite; standalone runs exit 0 —
    # gpurun_out/r2_call4.log, r2_call6.log), so a completed step8 with
    # a hung exit still passes, and the artifact is recorded here.
    assert "step8 capture+replay a2a/allreduce ok" in out, \
        f"probe rc={rc}; output:\n{out[-4000:]}"
    if rc != 0:
        import warnings
        warnings.warn("RCCL graph probe verified all steps but its exit "
                      f"hung under the pytest parent (rc={rc})")


def _gpu_round(rank, world, port, gather, out_q, seed=5):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": "0",
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "BLADES_AMD_BACKEND": "gloo",
    })
    from blades_amd import Simulator
    from blades_amd.datasets import SyntheticFLDataset
    from blades_amd.models import MLP

    ds = SyntheticFLDataset(num_clients=6, samples_per_client=16,
                            batch_size=8, shape=(1, 28, 28), num_classes=10,
                            seed=0, device="cuda:0")
    sim = Simulator(ds, num_byzantine=2, attack="alie",
                    attack_kws={"num_clients": 6, "num_byzantine": 2},
                    aggregator="trimmedmean", aggregator_kws={"nb": 2},
                    use_cuda=True, device="cuda:0", gather=gather,
                    log_path=f"/tmp/bl_mrgpu_{gather}_{world}_{rank}",
                    seed=seed)
    torch.manual_seed(seed)
    sim.run(MLP(), global_rounds=3, validate_interval=0, client_lr=0.1,
            server_lr=1.0)
    out_q.put((rank, sim.server.flat_parameters().cpu().numpy().copy()))
    if world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


def _spawn_world(world, port, gather):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_gpu_round, args=(r, world, port, gather, q))
             for r in range(world)]
    for p in procs:
        p.start()
    thetas = {}
    for _ in range(world):
        rank, theta = q.get(timeout=400)
        thetas[rank] = torch.from_numpy(theta)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    return thetas


@pytest.mark.timeout(900)
@pytest.mark.parametrize("gather", ["full", "coordinate"])
def test_two_process_gpu_round_matches_single_rank(gather):
    theta1 = _spawn_world(1, 29668, "full")[0]
    thetas = _spawn_world(2, 29669, gather)
    assert torch.equal(thetas[0], thetas[1]), "ranks diverged"
    assert torch.allclose(theta1, thetas[0], atol=1e-6), \
        (theta1 - thetas[0]).abs().max().item()
