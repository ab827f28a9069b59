"""End-to-end robustness at CIFAR scale: with 20%
ALIE attackers on learnable template data, TrimmedMean keeps the accuracy
that plain Mean loses.  The full slow curve (paper z, 300 rounds) lives in
scripts/robustness_curve.py + docs; this test uses an aggressive explicit
z so the separation appears within ~40 rounds."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def _train(aggregator, rounds=40, z=4.0, seed=7):
    from blades_amd import Simulator
    from blades_amd.datasets import SyntheticFLDataset
    from blades_amd.models import resnet18

    ds = SyntheticFLDataset(num_clients=100, samples_per_client=64,
                            batch_size=32, shape=(3, 32, 32), num_classes=10,
                            seed=0, device="cuda:0", learnable="templates")
    agg_kws = {"nb": 20} if aggregator == "trimmedmean" else {}
    sim = Simulator(ds, num_byzantine=20, attack="alie",
                    attack_kws={"num_clients": 100, "num_byzantine": 20, "z": z},
                    aggregator=aggregator, aggregator_kws=agg_kws,
                    use_cuda=True, log_path=f"/tmp/robust_t_{aggregator}",
                    seed=seed)
    torch.manual_seed(seed)
    sim.run(resnet18(norm="batch-local"), global_rounds=rounds, local_steps=1,
            validate_interval=0, client_lr=0.05, server_lr=1.0)
    _, top1 = sim.test_actor(rounds, batch_size=64)
    return float(top1)


@pytest.mark.timeout(900)
def test_trimmedmean_recovers_what_mean_loses_under_alie():
    top1_tm = _train("trimmedmean")
    top1_mean = _train("mean")
    # template data is easy: a robust aggregator should be well into
    # learning by round 40 while the attacked mean lags behind
    assert top1_tm > 0.6, f"trimmedmean failed to learn: {top1_tm}"
    assert top1_tm > top1_mean + 0.15, (top1_tm, top1_mean)
