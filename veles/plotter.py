"""Plotting units base + headless rendering.

Parity target: reference ``veles/plotter.py`` / ``veles/plotting_units.py``
(``Plotter`` base, ``AccumulatingPlotter``, ``MatrixPlotter``,
``ImagePlotter``, ``Histogram``) and the zmq GraphicsServer transport.

MI355X redesign: no live zmq/matplotlib-webagg client — plot units render
headless PNGs (matplotlib Agg) into ``root.common.dirs.plots`` and append
their series to a JSONL metrics sink, so training never blocks on a
display (reference goal) while CI can assert on the sink.
"""

import json
import os
import time

import numpy

from veles.config import root
from veles.units import Unit


class GraphicsServer(object):
    """Headless stand-in for the reference's zmq PUB graphics server:
    collects rendered artifacts, owns the output directory."""

    _instance = None

    def __init__(self, directory=None):
        self.directory = directory or root.common.dirs.get(
            "plots", os.path.join(root.common.dirs.get("base", "."),
                                  "plots"))
        os.makedirs(self.directory, exist_ok=True)
        self.metrics_path = os.path.join(self.directory, "metrics.jsonl")

    @classmethod
    def instance(cls):
        if cls._instance is None:
            cls._instance = cls()
        return cls._instance

    def emit(self, record):
        with open(self.metrics_path, "a") as f:
            f.write(json.dumps(record) + "\n")

    def save_figure(self, fig, name):
        path = os.path.join(self.directory, "%s.png" % name)
        fig.savefig(path, dpi=80)
        return path


class Plotter(Unit):
    """Base plotter: subclasses fill ``render(ax)``; the unit rate-limits
    redraws and never blocks the training loop (reference behavior via a
    separate process; here via cheap headless writes)."""

    def __init__(self, workflow, **kwargs):
        super(Plotter, self).__init__(workflow, **kwargs)
        self.plot_name = kwargs.get("name", type(self).__name__)
        self.redraw_interval = kwargs.get("redraw_interval", 1)
        self._runs = 0
        self.last_figure_path = None

    def run(self):
        self._runs += 1
        server = GraphicsServer.instance()
        record = self.record()
        if record is not None:
            record["plotter"] = self.plot_name
            record["t"] = time.time()
            server.emit(record)
        if self._runs % self.redraw_interval:
            return
        try:
            import matplotlib
            matplotlib.use("Agg", force=False)
            import matplotlib.pyplot as plt
        except Exception:
            return                       # sink-only when mpl is absent
        fig = plt.figure(figsize=(6, 4))
        ax = fig.add_subplot(111)
        try:
            self.render(ax)
            self.last_figure_path = server.save_figure(
                fig, self.plot_name.replace(" ", "_"))
        finally:
            plt.close(fig)

    # -- overridables --------------------------------------------------------
    def record(self):
        """Return a JSON-serializable dict for the metrics sink."""
        return None

    def render(self, ax):
        raise NotImplementedError


class AccumulatingPlotter(Plotter):
    """Accumulates scalar values (e.g. error per epoch) into a line."""

    def __init__(self, workflow, **kwargs):
        super(AccumulatingPlotter, self).__init__(workflow, **kwargs)
        self.input = None               # linked object
        self.input_field = kwargs.get("input_field", None)
        self.label = kwargs.get("label", self.plot_name)
        self.values = []

    def _value(self):
        v = self.input
        if self.input_field is not None:
            v = getattr(v, self.input_field) if not isinstance(
                v, (list, tuple, dict)) else v[self.input_field]
        if hasattr(v, "mem"):
            v.map_read()
            v = float(v.mem.reshape(-1)[0])
        return float(v)

    def record(self):
        self.values.append(self._value())
        return {"label": self.label, "value": self.values[-1],
                "n": len(self.values)}

    def render(self, ax):
        ax.plot(self.values, marker="o", markersize=2)
        ax.set_title(self.label)
        ax.set_xlabel("epoch")
        ax.grid(True)


class MatrixPlotter(Plotter):
    """Confusion-matrix style heatmap."""

    def __init__(self, workflow, **kwargs):
        super(MatrixPlotter, self).__init__(workflow, **kwargs)
        self.input = None               # linked 2-D Array / ndarray

    def _matrix(self):
        m = self.input
        if hasattr(m, "mem"):
            m.map_read()
            m = m.mem
        return numpy.asarray(m)

    def record(self):
        m = self._matrix()
        return {"shape": list(m.shape),
                "trace": float(numpy.trace(m)) if m.ndim == 2 else None,
                "sum": float(m.sum())}

    def render(self, ax):
        m = self._matrix()
        im = ax.imshow(m, cmap="viridis", interpolation="nearest")
        ax.figure.colorbar(im, ax=ax)
        ax.set_title(self.plot_name)


class ImagePlotter(Plotter):
    """Shows a sample image (first of the linked batch)."""

    def __init__(self, workflow, **kwargs):
        super(ImagePlotter, self).__init__(workflow, **kwargs)
        self.input = None

    def render(self, ax):
        img = self.input
        if hasattr(img, "mem"):
            img.map_read()
            img = img.mem
        img = numpy.asarray(img)
        if img.ndim == 4:
            img = img[0]
        if img.ndim == 3 and img.shape[-1] not in (1, 3):
            img = img[..., 0]
        ax.imshow(img.squeeze(), cmap="gray")
        ax.set_title(self.plot_name)


class Histogram(Plotter):
    def __init__(self, workflow, **kwargs):
        super(Histogram, self).__init__(workflow, **kwargs)
        self.input = None
        self.n_bars = kwargs.get("n_bars", 30)

    def render(self, ax):
        v = self.input
        if hasattr(v, "mem"):
            v.map_read()
            v = v.mem
        ax.hist(numpy.asarray(v).reshape(-1), bins=self.n_bars)
        ax.set_title(self.plot_name)
