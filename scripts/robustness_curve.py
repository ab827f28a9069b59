"""End-to-end robustness demonstration at CIFAR scale.

Trains ResNet-18 over 100 federated clients (20 ALIE attackers) on
teacher-labeled synthetic CIFAR-shaped data (template classes — learnable,
no network/dataset downloads) and compares aggregators: plain Mean loses
accuracy to the attack that TrimmedMean / Median recover — the simulator's
reason to exist, previously only unit-tested per formula.  The reference's
analogous check is the 2-D Gaussian example
(reference: examples/plot_comparing_aggregation_schemes.py:21-58).

Usage:  python scripts/robustness_curve.py [--rounds 300] [--z none|<float>]
Writes one JSON line per (aggregator, eval round) to stdout and a summary
markdown table at the end.
"""
import argparse
import json
import sys
import os

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def run_one(aggregator, rounds, z, seed=7, clients=100, byz=20,
            eval_every=25, attack="alie", eps=None):
    from blades_amd import Simulator
    from blades_amd.datasets import SyntheticFLDataset
    from blades_amd.models import resnet18

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    ds = SyntheticFLDataset(num_clients=clients, samples_per_client=64,
                            batch_size=32, shape=(3, 32, 32), num_classes=10,
                            seed=0, device=dev, learnable="templates")
    if attack == "alie":
        attack_kws = {"num_clients": clients, "num_byzantine": byz}
        if z is not None:
            attack_kws["z"] = z
    elif attack == "ipm":
        attack_kws = {"epsilon": 10.0 if eps is None else eps}
    elif attack == "noise":
        attack_kws = {"mean": 0.1, "std": 10.0}
    elif attack == "labelflipping":
        attack_kws = {"num_classes": 10}
    else:
        attack_kws = {}
    agg_kws = {}
    if aggregator == "trimmedmean":
        agg_kws = {"nb": byz}
    sim = Simulator(ds, num_byzantine=byz, attack=attack,
                    attack_kws=attack_kws,
                    aggregator=aggregator, aggregator_kws=agg_kws,
                    use_cuda=dev != "cpu", device=dev,
                    log_path=f"/tmp/robust_{attack}_{aggregator}", seed=seed)
    model = resnet18(norm="batch-local")
    torch.manual_seed(seed)
    curve = []
    sim.run(model, global_rounds=0, local_steps=1, validate_interval=0,
            client_lr=0.05, server_lr=1.0)
    clients_l = sim.get_clients()
    for r in range(1, rounds + 1):
        sim.train_round(r, 1, clients_l, 0.05)
        if r % eval_every == 0 or r == rounds:
            loss, top1 = sim.test_actor(r, batch_size=64)
            curve.append({"aggregator": aggregator, "round": r,
                          "top1": float(top1), "loss": float(loss)})
            print(json.dumps(curve[-1]), flush=True)
    return curve


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rounds", type=int, default=300)
    ap.add_argument("--eval-every", type=int, default=25)
    ap.add_argument("--z", type=str, default="none",
                    help="explicit ALIE z (default: paper z_max from n,m)")
    ap.add_argument("--aggregators", type=str,
                    default="mean,trimmedmean,median")
    ap.add_argument("--attack", type=str, default="alie",
                    choices=["alie", "ipm", "noise", "signflipping",
                             "labelflipping"])
    ap.add_argument("--eps", type=float, default=None, help="IPM epsilon")
    args = ap.parse_args()
    z = None if args.z == "none" else float(args.z)

    finals = {}
    curves = {}
    for agg in args.aggregators.split(","):
        c = run_one(agg, args.rounds, z, eval_every=args.eval_every,
                    attack=args.attack, eps=args.eps)
        curves[agg] = c
        finals[agg] = c[-1]["top1"]

    print("\n| aggregator | final top1 @ round %d (%s) |"
          % (args.rounds, args.attack))
    print("|---|---|")
    for agg, t in finals.items():
        print(f"| {agg} | {t:.3f} |")
    print(json.dumps({"final_top1": finals,
                      "curves": curves}), flush=True)


if __name__ == "__main__":
    main()
