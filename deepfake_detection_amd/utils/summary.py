"""Training summary CSV + optional per-metric plots
(reference dfd/timm/utils.py:205-248). matplotlib is optional in this
image; plots are skipped when it is absent."""

import csv
import os
from collections import OrderedDict


def update_summary(epoch, train_metrics, eval_metrics, filename, write_header=False,
                   plot=False):
    rowd = OrderedDict(epoch=epoch)
    rowd.update([("train_" + k, v) for k, v in train_metrics.items()])
    rowd.update([("eval_" + k, v) for k, v in eval_metrics.items()])
    with open(filename, mode="a") as cf:
        dw = csv.DictWriter(cf, fieldnames=rowd.keys())
        if write_header:
            dw.writeheader()
        dw.writerow(rowd)
    if plot:
        plot_csv(filename)


def plot_csv(filename):
    """Regenerate per-column plots next to the CSV (best-effort)."""
    try:
        import matplotlib

        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
        import pandas as pd
    except ImportError:
        return
    df = pd.read_csv(filename)
    plot_dir = os.path.join(os.path.dirname(filename), "plots")
    os.makedirs(plot_dir, exist_ok=True)
    for col in df.columns:
        if col == "epoch":
            continue
        fig = plt.figure()
        plt.plot(df["epoch"], df[col])
        plt.xlabel("epoch")
        plt.ylabel(col)
        fig.savefig(os.path.join(plot_dir, f"{col}.jpg"))
        plt.close(fig)
