"""Compare all built-in aggregators on 2-D synthetic data.

MI355X equivalent of reference examples/
plot_comparing_aggregation_schemes.py:21-58 — the dataset-free,
model-free aggregator check: 60 benign points around the origin, 40
outliers around (10, 10); robust schemes should land near the benign mean.
Writes a scatter plot when matplotlib is available, always prints the
aggregate each scheme produced.
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from blades_amd.aggregators import (Autogm, Centeredclipping,
                                    Clippedclustering, Clustering, Geomed,
                                    Krum, Mean, Median, Trimmedmean)

torch.manual_seed(0)
benign = torch.randn(60, 2)
outliers = torch.randn(40, 2) + 10.0
U = torch.cat([benign, outliers])

schemes = {
    "Mean": Mean(),
    "Median": Median(),
    "TrimmedMean": Trimmedmean(nb=40),
    "Krum": Krum(num_clients=100, num_byzantine=40),
    "GeoMed": Geomed(),
    "AutoGM": Autogm(lamb=2.0),
    "CenteredClipping": Centeredclipping(tau=10.0),
    "Clustering": Clustering(),
    "ClippedClustering": Clippedclustering(),
}

results = {}
for name, agg in schemes.items():
    out = agg(U.clone())
    results[name] = out
    print(f"{name:>18}: ({out[0]:+.3f}, {out[1]:+.3f})")

try:
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    fig, ax = plt.subplots(figsize=(7, 7))
    ax.scatter(benign[:, 0], benign[:, 1], s=12, alpha=0.5, label="benign")
    ax.scatter(outliers[:, 0], outliers[:, 1], s=12, alpha=0.5,
               label="byzantine")
    for name, out in results.items():
        ax.scatter([out[0]], [out[1]], marker="x", s=120, label=name)
    ax.legend(fontsize=8)
    fig.savefig("aggregation_schemes.png", dpi=120)
    print("wrote aggregation_schemes.png")
except ImportError:
    print("(matplotlib not installed; skipping the plot)")
