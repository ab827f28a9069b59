"""Fused many-model engine vs reference-semantics loop engine parity."""
import pytest
import torch

from blades_amd import Simulator
from blades_amd.datasets import SyntheticFLDataset
from blades_amd.engine import FusedEngine, ParamSpec, make_vmap_safe
from blades_amd.models import MLP, resnet18


def _run(engine, steps, attack=None, num_byz=0, model_fn=MLP, shape=(1, 28, 28),
         attack_kws=None):
    torch.manual_seed(7)
    ds = SyntheticFLDataset(num_clients=6, samples_per_client=16, batch_size=4,
                            shape=shape, num_classes=10, seed=0)
    sim = Simulator(ds, num_byzantine=num_byz, attack=attack,
                    attack_kws=attack_kws or {}, aggregator="mean",
                    log_path=f"/tmp/bl_engine_{engine}", seed=42, engine=engine)
    sim.run(model_fn(), global_rounds=2, local_steps=steps, client_lr=0.05,
            server_lr=1.0, validate_interval=0)
    return sim.server.flat_parameters()


@pytest.mark.parametrize("steps", [1, 3])
def test_fused_matches_loop_mlp(steps):
    a = _run("auto", steps)
    b = _run("loop", steps)
    assert torch.allclose(a, b, atol=1e-6)


@pytest.mark.parametrize("attack,kws", [
    ("labelflipping", {}),
    ("signflipping", {}),
])
def test_fused_matches_loop_training_time_attacks(attack, kws):
    a = _run("auto", 2, attack=attack, num_byz=2, attack_kws=kws)
    b = _run("loop", 2, attack=attack, num_byz=2, attack_kws=kws)
    assert torch.allclose(a, b, atol=1e-6)


def test_fused_matches_loop_resnet_batchstats():
    """ResNet-18 with batch-stats norm: fused grouped path == per-client loop."""
    fn = lambda: resnet18(norm="batch-local")
    a = _run("auto", 1, model_fn=fn, shape=(3, 32, 32))
    b = _run("loop", 1, model_fn=fn, shape=(3, 32, 32))
    # vmapped (grouped) conv backward reduces in a different order than the
    # per-client plain conv backward — fp32 accumulation noise only
    assert torch.allclose(a, b, atol=5e-4)
    assert (a - b).abs().mean() < 1e-6


def test_make_vmap_safe_strips_running_stats():
    m = resnet18(norm="batch")
    make_vmap_safe(m)
    for mod in m.modules():
        if isinstance(mod, torch.nn.BatchNorm2d):
            assert mod.running_mean is None and not mod.track_running_stats


def test_update_semantics_theta_after_minus_before():
    """update = θ_after − θ_before (reference: client.py:130,216-228)."""
    torch.manual_seed(0)
    model = MLP()
    spec = ParamSpec.from_module(model)
    theta = spec.flatten(model)
    eng = FusedEngine(model, spec, torch.device("cpu"))
    ds = SyntheticFLDataset(num_clients=3, samples_per_client=8, batch_size=4,
                            shape=(1, 28, 28), num_classes=10, seed=0)
    from blades_amd.client import BladesClient
    clients = [BladesClient(id=i) for i in range(3)]
    U = eng.run_round(theta, clients, ds, local_steps=2, lr=0.1)
    assert U.shape == (3, spec.d)
    # manual per-client SGD replay for client 0
    ds2 = SyntheticFLDataset(num_clients=3, samples_per_client=8, batch_size=4,
                             shape=(1, 28, 28), num_classes=10, seed=0)
    import copy
    m = copy.deepcopy(model)
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    for x, y in ds2.get_train_data(0, 2):
        opt.zero_grad()
        loss = torch.clamp(torch.nn.functional.cross_entropy(m(x), y), 0, 1e6)
        loss.backward()
        opt.step()
    manual = spec.flatten(m) - theta
    assert torch.allclose(U[0], manual, atol=1e-6)


def test_client_chunking_equivalent():
    torch.manual_seed(0)
    model = MLP()
    spec = ParamSpec.from_module(model)
    theta = spec.flatten(model)
    from blades_amd.client import BladesClient
    clients = [BladesClient(id=i) for i in range(5)]
    outs = []
    for chunk in (None, 2):
        ds = SyntheticFLDataset(num_clients=5, samples_per_client=8,
                                batch_size=4, shape=(1, 28, 28),
                                num_classes=10, seed=0)
        eng = FusedEngine(model, spec, torch.device("cpu"), client_chunk=chunk)
        outs.append(eng.run_round(theta.clone(), clients, ds, 1, 0.1))
    assert torch.allclose(outs[0], outs[1], atol=1e-7)


def test_stacked_eval_matches_per_client_eval():
    """Batched eval fast path == per-client DataLoader eval."""
    import copy

    from blades_amd.datasets import SyntheticFLDataset
    from blades_amd.client import BladesClient
    from blades_amd.engine import FusedEngine, LoopEngine, ParamSpec
    from blades_amd.utils import top1_accuracy

    torch.manual_seed(1)
    model = MLP()
    spec = ParamSpec.from_module(model)
    theta = spec.flatten(model)
    ds = SyntheticFLDataset(num_clients=5, samples_per_client=8, batch_size=4,
                            shape=(1, 28, 28), num_classes=10, seed=0,
                            test_samples_per_client=12)
    clients = [BladesClient(id=i) for i in range(5)]
    eng = FusedEngine(model, spec, torch.device("cpu"))
    fast = eng.evaluate(theta, clients, ds, 1, 4, {"top1": top1_accuracy})

    # reference path: loop engine with per-client model
    for c in clients:
        c.set_model(model)
    loop = LoopEngine(device="cpu")
    slow = loop.evaluate(model, clients, ds, 1, 4, {"top1": top1_accuracy})

    assert len(fast) == len(slow) == 5
    for f, s in zip(fast, slow):
        assert f["Length"] == s["Length"]
        assert abs(f["Loss"] - s["Loss"]) < 1e-5, (f, s)
        assert abs(f["top1"] - s["top1"]) < 1e-3
