"""GPU integration: fused engine + full simulator rounds on MI355X."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_fused_engine_resnet_round_gpu():
    from blades_amd import Simulator
    from blades_amd.datasets import SyntheticFLDataset
    from blades_amd.models import resnet18

    ds = SyntheticFLDataset(num_clients=16, samples_per_client=32,
                            batch_size=16, shape=(3, 32, 32), num_classes=10,
                            seed=0, device="cuda:0")
    sim = Simulator(ds, num_byzantine=3, attack="alie",
                    attack_kws={"num_clients": 16, "num_byzantine": 3},
                    aggregator="trimmedmean", aggregator_kws={"nb": 3},
                    use_cuda=True, log_path="/tmp/bl_gpu_int", seed=0)
    ret = sim.run(resnet18(norm="batch-local"), global_rounds=2,
                  local_steps=1, client_lr=0.1, server_lr=1.0,
                  validate_interval=2, test_batch_size=32)
    assert len(ret) == 2
    theta = sim.server.flat_parameters()
    assert torch.isfinite(theta).all()


def test_gpu_matches_cpu_run():
    """Same seed, same data: the GPU (HIP kernels + fused engine) run must
    match the CPU run to fp32 accumulation tolerance."""
    from blades_amd import Simulator
    from blades_amd.datasets import SyntheticFLDataset
    from blades_amd.models import MLP

    def go(use_cuda):
        ds = SyntheticFLDataset(num_clients=8, samples_per_client=16,
                                batch_size=8, shape=(1, 28, 28),
                                num_classes=10, seed=0,
                                device="cuda:0" if use_cuda else "cpu")
        # mean, not median: selection aggregators turn ~1e-7 cross-device
        # fp32 noise into discrete order-statistic swaps, which no tolerance
        # cleanly bounds; continuous aggregators keep the drift linear
        sim = Simulator(ds, num_byzantine=2, attack="ipm",
                        aggregator="mean", use_cuda=use_cuda,
                        log_path=f"/tmp/bl_xdev_{use_cuda}", seed=7)
        sim.run(MLP(), global_rounds=3, local_steps=2, client_lr=0.1,
                server_lr=1.0, validate_interval=0)
        return sim.server.flat_parameters().cpu()

    a = go(True)
    b = go(False)
    assert torch.allclose(a, b, atol=5e-4)
    assert (a - b).abs().mean() < 1e-5


def test_all_aggregators_run_gpu():
    from blades_amd.aggregators import get_aggregator
    from blades_amd.client import BladesClient

    g = torch.Generator(device="cuda")
    g.manual_seed(0)
    U = torch.randn(20, 50000, generator=g, device="cuda")
    clients = []
    for i in range(20):
        c = BladesClient(id=i, device="cuda:0")
        c.save_update(U[i])
        clients.append(c)
    clients[0].trust()

    for name, kws in [
        ("mean", {}), ("median", {}), ("trimmedmean", {"nb": 5}),
        ("krum", {"num_clients": 20, "num_byzantine": 5}),
        ("multikrum", {"num_clients": 20, "num_byzantine": 5, "m": 3}),
        ("geomed", {}), ("autogm", {"lamb": 1.0}),
        ("centeredclipping", {}), ("clustering", {}),
        ("clippedclustering", {}), ("fltrust", {}),
    ]:
        agg = get_aggregator(name, **kws)
        out = agg(clients)
        assert out.shape == (50000,), name
        assert torch.isfinite(out).all(), name


def test_graph_round_matches_eager():
    """hipGraph-captured rounds must reproduce the eager rounds exactly
    (same kernels, same order) on ALIE+TrimmedMean."""
    from blades_amd import Simulator
    from blades_amd.datasets import SyntheticFLDataset
    from blades_amd.models import resnet18

    def go(graphs):
        ds = SyntheticFLDataset(num_clients=12, samples_per_client=32,
                                batch_size=8, shape=(3, 32, 32),
                                num_classes=10, seed=0, device="cuda:0")
        sim = Simulator(ds, num_byzantine=3, attack="alie",
                        attack_kws={"num_clients": 12, "num_byzantine": 3},
                        aggregator="trimmedmean", aggregator_kws={"nb": 3},
                        use_cuda=True, log_path=f"/tmp/bl_graph_{graphs}",
                        seed=3, hip_graphs=graphs)
        sim.run(resnet18(norm="batch-local"), global_rounds=5, local_steps=1,
                client_lr=0.1, server_lr=1.0, validate_interval=0)
        used_graph = sim._graph_round is not None and sim._graph_round.graph is not None
        return sim.server.flat_parameters().cpu(), used_graph

    tg, used = go(True)
    te, _ = go(False)
    assert used, "graph capture did not engage"
    assert torch.allclose(tg, te, atol=2e-5), (tg - te).abs().max()


def test_scheduler_under_graphs():
    """LR schedulers must work across graph replays (learning rates live in
    device scalars; no re-capture)."""
    from blades_amd import Simulator
    from blades_amd.datasets import SyntheticFLDataset
    from blades_amd.models import resnet18

    def go(graphs):
        ds = SyntheticFLDataset(num_clients=8, samples_per_client=16,
                                batch_size=8, shape=(3, 32, 32),
                                num_classes=10, seed=0, device="cuda:0")
        sim = Simulator(ds, num_byzantine=2, attack="alie",
                        attack_kws={"num_clients": 8, "num_byzantine": 2},
                        aggregator="trimmedmean", aggregator_kws={"nb": 2},
                        use_cuda=True, log_path=f"/tmp/bl_sched_{graphs}",
                        seed=2, hip_graphs=graphs)
        model = resnet18(norm="batch-local")
        server_opt = None
        # client scheduler halves lr at rounds 3 and 5
        class FakeSched:
            def __init__(self):
                self.lr = 0.02
                self.n = 0
            def step(self):
                self.n += 1
                if self.n in (3, 5):
                    self.lr /= 2
            def get_last_lr(self):
                return [self.lr]
        ret = sim.run(model, global_rounds=6, local_steps=1, client_lr=0.02,
                      server_lr=1.0, validate_interval=0,
                      client_lr_scheduler=FakeSched())
        used = sim._graph_round is not None and sim._graph_round.graph is not None
        return sim.server.flat_parameters().cpu(), used

    tg, used = go(True)
    te, _ = go(False)
    assert used
    assert torch.allclose(tg, te, atol=5e-4), (tg - te).abs().max()


def test_mixed_population_gpu():
    """Custom client subclass (loop engine) + fused rest on GPU."""
    from blades_amd import BladesClient, Simulator
    from blades_amd.datasets import SyntheticFLDataset
    from blades_amd.models import MLP

    seen = []

    class MyClient(BladesClient):
        def local_training(self, data_batches):
            seen.append(self.id())
            super().local_training(data_batches)

    ds = SyntheticFLDataset(num_clients=6, samples_per_client=16, batch_size=8,
                            shape=(1, 28, 28), num_classes=10, seed=0,
                            device="cuda:0")
    sim = Simulator(ds, use_cuda=True, log_path="/tmp/bl_mixed_gpu", seed=0)
    c = MyClient(id=2, device="cuda:0")
    sim._clients[2] = c
    sim.run(MLP(), global_rounds=2, validate_interval=0)
    assert seen == [2, 2]
    assert torch.isfinite(sim.server.flat_parameters()).all()


def test_graph_round_fedavg_matches_eager():
    """hipGraph capture with local_steps > 1 (FedAvg slab path in-graph)."""
    from blades_amd import Simulator
    from blades_amd.datasets import SyntheticFLDataset
    from blades_amd.models import MLP

    def go(graphs):
        ds = SyntheticFLDataset(num_clients=8, samples_per_client=32,
                                batch_size=8, shape=(1, 28, 28),
                                num_classes=10, seed=0, device="cuda:0")
        sim = Simulator(ds, num_byzantine=2, attack="ipm",
                        aggregator="median", use_cuda=True,
                        log_path=f"/tmp/bl_gfa_{graphs}", seed=4,
                        hip_graphs=graphs)
        sim.run(MLP(), global_rounds=4, local_steps=3, client_lr=0.05,
                server_lr=1.0, validate_interval=0)
        used = sim._graph_round is not None and sim._graph_round.graph is not None
        return sim.server.flat_parameters().cpu(), used

    tg, used = go(True)
    te, _ = go(False)
    assert used
    assert torch.allclose(tg, te, atol=1e-5)
