import sys, torch
sys.path.insert(0, "/root/repo")
from blades_amd import Simulator
from blades_amd.datasets import SyntheticFLDataset
from blades_amd.models import resnet18

def run(tag):
    ds = SyntheticFLDataset(num_clients=16, samples_per_client=32, batch_size=8,
                            shape=(3, 32, 32), num_classes=10, seed=0,
                            device="cuda:0")
    sim = Simulator(ds, num_byzantine=3, attack="alie",
                    attack_kws={"num_clients": 16, "num_byzantine": 3},
                    aggregator="trimmedmean", aggregator_kws={"nb": 3},
                    use_cuda=True, log_path=f"/tmp/det_{tag}", seed=77)
    torch.manual_seed(77)
    sim.run(resnet18(norm="batch-local"), global_rounds=6, local_steps=1,
            client_lr=0.1, server_lr=1.0, validate_interval=0)
    return sim.server.flat_parameters().cpu()

a = run("a")
b = run("b")
print("bitwise equal across independent runs:", torch.equal(a, b))
print("max diff:", (a - b).abs().max().item())
