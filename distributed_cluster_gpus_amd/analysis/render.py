"""Rendering backend for the analysis layer.

Every figure is produced as DATA first (a DataFrame written to
``<out>/<name>.csv``); if matplotlib is importable, a PNG is rendered too.
This keeps the full analysis capability usable in headless images without
matplotlib (this ROCm image ships pandas but not matplotlib)."""
import os
from typing import Optional

import numpy as np
import pandas as pd


def have_mpl() -> bool:
    try:
        import matplotlib  # noqa: F401
        return True
    except ImportError:
        return False


def _plt():
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    return plt


def emit(df: pd.DataFrame, out_dir: str, name: str,
         kind: str = "line", x: Optional[str] = None,
         y=None, title: str = "", ylabel: str = "",
         hue: Optional[str] = None, logy: bool = False) -> str:
    """Write <name>.csv always; render <name>.png when matplotlib exists.
    kind: line | bar | hist | scatter | heatmap | box | violin | boxen."""
    os.makedirs(out_dir, exist_ok=True)
    csv_path = os.path.join(out_dir, f"{name}.csv")
    df.to_csv(csv_path, index=False)
    if not have_mpl() or len(df) == 0:
        return csv_path
    plt = _plt()
    fig, ax = plt.subplots(figsize=(9, 5))
    try:
        if kind == "line":
            if hue is not None:
                for key, sub in df.groupby(hue):
                    ax.plot(sub[x], sub[y], label=str(key))
                ax.legend(fontsize=8)
            else:
                for col in ([y] if isinstance(y, str) else y):
                    ax.plot(df[x], df[col], label=col)
                ax.legend(fontsize=8)
            ax.set_xlabel(x)
        elif kind == "bar":
            ax.bar(df[x].astype(str), df[y])
            ax.set_xlabel(x)
            ax.tick_params(axis="x", rotation=30)
        elif kind == "hist":
            if hue is not None:
                for key, sub in df.groupby(hue):
                    ax.hist(sub[y], bins=60, alpha=0.5, label=str(key))
                ax.legend(fontsize=8)
            else:
                ax.hist(df[y], bins=60)
            ax.set_xlabel(y)
        elif kind == "scatter":
            if hue is not None:
                for key, sub in df.groupby(hue):
                    ax.scatter(sub[x], sub[y], s=12, label=str(key))
                ax.legend(fontsize=8)
            else:
                ax.scatter(df[x], df[y], s=12)
            ax.set_xlabel(x)
        elif kind == "box":
            groups = [(str(k), sub[y].values) for k, sub in df.groupby(hue)
                      if len(sub)]
            if groups:
                ax.boxplot([g[1] for g in groups],
                           labels=[g[0] for g in groups], showfliers=False)
                ax.tick_params(axis="x", rotation=30)
        elif kind == "violin":
            groups = [(str(k), sub[y].dropna().values)
                      for k, sub in df.groupby(hue) if len(sub)]
            if groups:
                ax.violinplot([g[1] for g in groups], showmedians=True,
                              showextrema=False)
                ax.set_xticks(range(1, len(groups) + 1))
                ax.set_xticklabels([g[0] for g in groups], rotation=30)
        elif kind == "boxen":
            # letter-value plot: nested boxes at halving tail depths
            # (reference renders this via seaborn.boxenplot; same construct)
            from matplotlib.patches import Rectangle
            groups = [(str(k), np.sort(sub[y].dropna().values))
                      for k, sub in df.groupby(hue) if len(sub)]
            for i, (label, vals) in enumerate(groups, start=1):
                if len(vals) < 2:
                    continue
                depth, width = 0.25, 0.8
                while depth * len(vals) >= 1 and width > 0.05:
                    lo = float(np.quantile(vals, depth))
                    hi = float(np.quantile(vals, 1 - depth))
                    ax.add_patch(Rectangle((i - width / 2, lo), width,
                                           max(hi - lo, 1e-12),
                                           facecolor="C%d" % ((i - 1) % 10),
                                           alpha=0.35, edgecolor="none"))
                    depth /= 2
                    width *= 0.7
                med = float(np.median(vals))
                ax.plot([i - 0.4, i + 0.4], [med, med], color="k", lw=1.2)
            if groups:
                ax.set_xlim(0.4, len(groups) + 0.6)
                ax.set_xticks(range(1, len(groups) + 1))
                ax.set_xticklabels([g[0] for g in groups], rotation=30)
        elif kind == "heatmap":
            mat = df.set_index(df.columns[0])
            im = ax.imshow(mat.values, aspect="auto", cmap="viridis")
            ax.set_xticks(range(len(mat.columns)))
            ax.set_xticklabels(mat.columns, rotation=45, fontsize=7)
            ax.set_yticks(range(len(mat.index)))
            ax.set_yticklabels(mat.index, fontsize=7)
            fig.colorbar(im, ax=ax)
        if logy:
            ax.set_yscale("log")
        ax.set_title(title or name)
        if ylabel:
            ax.set_ylabel(ylabel)
        fig.tight_layout()
        fig.savefig(os.path.join(out_dir, f"{name}.png"), dpi=110)
    except Exception as e:  # rendering must never kill a report: the CSV
        print(f"[render] {name}.png skipped: {e}")   # data is the artifact
    finally:
        plt.close(fig)
    return csv_path
