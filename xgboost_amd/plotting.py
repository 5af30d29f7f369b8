"""Plotting utilities (reference: python-package/xgboost/plotting.py).
matplotlib/graphviz are optional; functions raise ImportError when the
backend library is unavailable."""
from __future__ import annotations

from typing import Optional

import numpy as np



def plot_importance(booster, ax=None, height: float = 0.2,
                    xlim=None, ylim=None, title: str = "Feature importance",
                    xlabel: str = "Importance score",
                    ylabel: str = "Features",
                    importance_type: str = "weight",
                    max_num_features: Optional[int] = None,
                    grid: bool = True, show_values: bool = True,
                    values_format: str = "{v}", **kwargs):
    try:
        import matplotlib.pyplot as plt
    except ImportError as e:
        raise ImportError("plot_importance requires matplotlib") from e
    b = booster.get_booster() if hasattr(booster, "get_booster") else booster
    if isinstance(b, dict):
        importance = b
    else:
        importance = b.get_score(importance_type=importance_type)
    if not importance:
        raise ValueError("Booster is empty")
    tuples = sorted(importance.items(), key=lambda x: x[1])
    if max_num_features is not None:
        tuples = tuples[-max_num_features:]
    labels, values = zip(*tuples)
    if ax is None:
        _, ax = plt.subplots(1, 1)
    ylocs = np.arange(len(values))
    ax.barh(ylocs, values, align="center", height=height, **kwargs)
    if show_values:
        for x, y in zip(values, ylocs):
            ax.text(x + 1, y, values_format.format(v=x), va="center")
    ax.set_yticks(ylocs)
    ax.set_yticklabels(labels)
    if xlim is not None:
        ax.set_xlim(xlim)
    if ylim is not None:
        ax.set_ylim(ylim)
    if title:
        ax.set_title(title)
    if xlabel:
        ax.set_xlabel(xlabel)
    if ylabel:
        ax.set_ylabel(ylabel)
    ax.grid(grid)
    return ax


def to_graphviz(booster, num_trees: int = 0, rankdir: Optional[str] = None,
                yes_color: Optional[str] = None,
                no_color: Optional[str] = None,
                condition_node_params: Optional[dict] = None,
                leaf_node_params: Optional[dict] = None, **kwargs):
    try:
        from graphviz import Source
    except ImportError as e:
        raise ImportError("to_graphviz requires the graphviz package") from e
    b = booster.get_booster() if hasattr(booster, "get_booster") else booster
    dot = b.get_dump(dump_format="dot")[num_trees]
    return Source(dot)


def plot_tree(booster, num_trees: int = 0, rankdir: Optional[str] = None,
              ax=None, **kwargs):
    try:
        import matplotlib.pyplot as plt
        import matplotlib.image as image
    except ImportError as e:
        raise ImportError("plot_tree requires matplotlib") from e
    from io import BytesIO
    g = to_graphviz(booster, num_trees=num_trees, rankdir=rankdir, **kwargs)
    s = BytesIO(g.pipe(format="png"))
    img = image.imread(s)
    if ax is None:
        _, ax = plt.subplots(1, 1)
    ax.imshow(img)
    ax.axis("off")
    return ax
