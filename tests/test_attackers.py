"""Attacker semantics tests (SURVEY.md §2.5 formulas)."""
import math

import pytest
import torch
from scipy.stats import norm as scipy_norm

from blades_amd import Simulator
from blades_amd.attackers import (AlieClient, IpmClient, LabelflippingClient,
                                  NoiseClient, SignflippingClient,
                                  get_attacker_cls)
from blades_amd.datasets import SyntheticFLDataset
from blades_amd.models import MLP


def make_sim(attack, num_byz=3, K=8, attack_kws=None, engine="auto", seed=42):
    ds = SyntheticFLDataset(num_clients=K, samples_per_client=16, batch_size=8,
                            shape=(1, 28, 28), num_classes=10, seed=0)
    return Simulator(ds, num_byzantine=num_byz, attack=attack,
                     attack_kws=attack_kws or {}, aggregator="mean",
                     log_path="/tmp/bl_attack_test", seed=seed, engine=engine)


def run_one_round(sim, steps=1):
    sim.run(MLP(), global_rounds=1, local_steps=steps, client_lr=0.1,
            server_lr=1.0, validate_interval=0)
    return torch.stack([c.get_update() for c in sim.get_clients()])


def test_alie_formula():
    K, f = 8, 3
    sim = make_sim("alie", num_byz=f, K=K,
                   attack_kws={"num_clients": K, "num_byzantine": f})
    U = run_one_round(sim)
    honest = U[f:]
    mu = honest.mean(0)
    std = honest.std(0)  # Bessel-corrected, as the reference's torch.std
    s = math.floor(K / 2 + 1) - f
    z = scipy_norm.ppf((K - f - s) / (K - f))
    expected = mu - std * z
    for i in range(f):
        assert torch.allclose(U[i], expected, atol=1e-5)


def test_alie_explicit_z():
    c = AlieClient(num_clients=10, num_byzantine=2, z=1.5)
    assert c.z_max == 1.5


def test_ipm_formula():
    K, f = 8, 2
    sim = make_sim("ipm", num_byz=f, K=K, attack_kws={"epsilon": 0.5})
    U = run_one_round(sim)
    expected = -0.5 * U[f:].mean(0)
    for i in range(f):
        assert torch.allclose(U[i], expected, atol=1e-5)


def test_noise_is_deterministic_per_round():
    sim1 = make_sim("noise", num_byz=2)
    U1 = run_one_round(sim1)
    sim2 = make_sim("noise", num_byz=2)
    U2 = run_one_round(sim2)
    # same seed -> identical noise; distribution roughly N(0.1, 0.1)
    assert torch.allclose(U1[0], U2[0])
    assert abs(U1[0].mean().item() - 0.1) < 0.01
    assert not torch.allclose(U1[0], U1[1])  # distinct per-client streams


def test_labelflipping_transform():
    c = LabelflippingClient(num_classes=10)
    y = torch.tensor([0, 3, 9])
    _, y2 = c.on_train_batch_begin(None, y)
    assert torch.equal(y2, torch.tensor([9, 6, 0]))
    assert torch.equal(c.fused_target_transform(y), y2)


def test_signflipping_is_exact_negation_at_one_step():
    """With 1 local step from shared θ and identical data, the sign-flipped
    update is exactly −(honest update) (gradient ascent semantics)."""
    U_byz = run_one_round(make_sim("signflipping", num_byz=2, K=8))
    U_hon = run_one_round(make_sim(None, num_byz=0, K=8))
    assert torch.allclose(U_byz[0], -U_hon[0], atol=1e-6)
    assert torch.allclose(U_byz[1], -U_hon[1], atol=1e-6)
    # honest clients unaffected
    assert torch.allclose(U_byz[5], U_hon[5], atol=1e-6)


def test_registry_resolves_all():
    for name in ["alie", "ipm", "labelflipping", "noise", "signflipping"]:
        assert get_attacker_cls(name) is not None


def test_custom_attacker_via_register(tmp_path):
    """User-defined attacker through register_attackers (reference:
    examples/customize_attack.py path)."""
    from blades_amd.client import ByzantineClient

    class MaliciousClient(ByzantineClient):
        def omniscient_callback(self, simulator):
            cur = self.get_update()
            self._state["saved_update"] = torch.ones_like(cur) * 7.0

    ds = SyntheticFLDataset(num_clients=6, samples_per_client=16, batch_size=8,
                            shape=(1, 28, 28), num_classes=10, seed=0)
    sim = Simulator(ds, aggregator="mean", log_path=str(tmp_path), seed=1)
    sim.register_attackers([MaliciousClient(), MaliciousClient()])
    U = run_one_round(sim)
    assert torch.all(U[0] == 7.0) and torch.all(U[1] == 7.0)
    assert sim.get_clients()[0].is_byzantine()
