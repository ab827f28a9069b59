"""Checkpoint/resume: a resumed run must produce bit-identical training."""
import os

import torch

from blades_amd import Simulator
from blades_amd.checkpoint import load_checkpoint, save_checkpoint
from blades_amd.datasets import SyntheticFLDataset
from blades_amd.models import MLP


def make_sim(tmp, tag, seed=11):
    ds = SyntheticFLDataset(num_clients=6, samples_per_client=16, batch_size=8,
                            shape=(1, 28, 28), num_classes=10, seed=0)
    return Simulator(ds, num_byzantine=2, attack="ipm", aggregator="centeredclipping",
                     log_path=os.path.join(tmp, tag), seed=seed)


def test_flat_roundtrip(tmp_path):
    m = MLP()
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    path = str(tmp_path / "ck.pt")
    save_checkpoint(path, m, opt, round_idx=5)
    m2 = MLP()
    ck = load_checkpoint(path, m2, torch.optim.SGD(m2.parameters(), lr=0.1))
    assert ck["round"] == 5
    for p1, p2 in zip(m.parameters(), m2.parameters()):
        assert torch.equal(p1, p2)


def test_layout_mismatch_rejected(tmp_path):
    m = MLP()
    path = str(tmp_path / "ck.pt")
    save_checkpoint(path, m)
    from blades_amd.models import cct_2_3x2_32
    try:
        load_checkpoint(path, cct_2_3x2_32())
        assert False, "should have raised"
    except ValueError:
        pass


def test_resume_reproduces_stateful_aggregator(tmp_path):
    """Train 4 rounds straight vs 2 rounds + checkpoint + resume 2 rounds:
    identical final parameters (incl. Centeredclipping momentum state)."""
    tmp = str(tmp_path)

    sim_a = make_sim(tmp, "a")
    model_a = MLP()
    sim_a.run(model_a, global_rounds=4, validate_interval=0, client_lr=0.1)
    theta_a = sim_a.server.flat_parameters()

    sim_b = make_sim(tmp, "b")
    model_b = MLP()
    sim_b.run(model_b, global_rounds=2, validate_interval=0, client_lr=0.1)
    ck = str(tmp_path / "resume.pt")
    save_checkpoint(ck, model_b, sim_b.server_opt, round_idx=2,
                    aggregator=sim_b.aggregator)

    sim_c = make_sim(tmp, "c")
    model_c = MLP()
    # run() resets weights, so start it for 0 rounds to build the server,
    # then load the checkpoint and continue
    sim_c.run(model_c, global_rounds=0, validate_interval=0, client_lr=0.1)
    load_checkpoint(ck, model_c, sim_c.server_opt, aggregator=sim_c.aggregator)
    for r in (3, 4):
        sim_c.train_round(r, 1, sim_c.get_clients(), 0.1)
    theta_c = sim_c.server.flat_parameters()

    assert torch.allclose(theta_a, theta_c, atol=1e-7)
