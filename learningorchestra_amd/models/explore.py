"""Explore-verb plotting — the native replacement for the reference's
seaborn-scatterplot-to-PNG path (database_executor_image/utils.py:300-309).

Drive it reflectively through the explore/{tool} endpoints::

    POST /explore/torch {"name": "plot1",
        "modulePath": "learningorchestra_amd.models.explore", "class": "Plot",
        "classParameters": {}, "method": "scatter",
        "methodParameters": {"data": "$titanic", "x": "Age", "y": "Fare"}}

The returned matplotlib Figure is rendered to PNG by the executor's result
store; ``GET /explore/{tool}/plot1`` serves the image.
"""
from __future__ import annotations

from typing import Optional


class Plot:
    """Matplotlib-backed plotting surface (seaborn is not in the image)."""

    def __init__(self, figsize=(8, 6), dpi: int = 100, style: Optional[str] = None):
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
        self._plt = plt
        self.figsize = tuple(figsize)
        self.dpi = dpi

    def _fig(self, title, xlabel, ylabel):
        fig, ax = self._plt.subplots(figsize=self.figsize, dpi=self.dpi)
        if title:
            ax.set_title(title)
        if xlabel:
            ax.set_xlabel(xlabel)
        if ylabel:
            ax.set_ylabel(ylabel)
        return fig, ax

    def scatter(self, data, x: str, y: str, hue: Optional[str] = None,
                title: str = ""):
        fig, ax = self._fig(title, x, y)
        if hue and hue in data:
            for val, grp in data.groupby(hue):
                ax.scatter(grp[x], grp[y], label=str(val), s=12, alpha=0.7)
            ax.legend(title=hue)
        else:
            ax.scatter(data[x], data[y], s=12, alpha=0.7)
        return fig

    def histogram(self, data, x: str, bins: int = 30, title: str = ""):
        fig, ax = self._fig(title, x, "count")
        ax.hist(data[x].dropna(), bins=bins)
        return fig

    def line(self, data, x: str, y: str, title: str = ""):
        fig, ax = self._fig(title, x, y)
        d = data.sort_values(x)
        ax.plot(d[x], d[y])
        return fig

    def bar(self, data, x: str, y: str, title: str = ""):
        fig, ax = self._fig(title, x, y)
        ax.bar(data[x].astype(str), data[y])
        return fig
