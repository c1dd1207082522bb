"""NN-specific plotting units.

Parity target: reference ``znicz/nn_plotting_units.py`` (``Weights2D``
kernel image grids, ``MSEHistogram``, ``KohonenHits``,
``KohonenNeighborMap``) plus the error-per-epoch accumulating plotter
the samples wire up.  Headless rendering (see veles/plotter.py).
"""

import numpy

from veles.plotter import AccumulatingPlotter, Histogram, Plotter


class ErrorPlotter(AccumulatingPlotter):
    """Error %% per epoch for one class (TRAIN/VALID/TEST)."""

    def __init__(self, workflow, **kwargs):
        kwargs.setdefault("name", "error_pt")
        super(ErrorPlotter, self).__init__(workflow, **kwargs)
        # ``values`` is linked to decision.epoch_n_err_pt by
        # link_error_plotter (overrides the base's list)
        self.clazz = kwargs.get("clazz", 1)   # VALID by default

    def _value(self):
        return float(self.values[self.clazz])

    def record(self):
        v = self._value()
        self._accum = getattr(self, "_accum", [])
        self._accum.append(v)
        return {"label": self.label, "value": v, "n": len(self._accum)}

    def render(self, ax):
        ax.plot(getattr(self, "_accum", []), marker="o", markersize=2)
        ax.set_title("error %% (class %d)" % self.clazz)
        ax.grid(True)


class Weights2D(Plotter):
    """Grid of first-layer kernels as small images (reference
    Weights2D): works for conv [K, ky, kx, C] and FC [out, in] weights
    with a known ``color_space``/``reshape_to``."""

    def __init__(self, workflow, **kwargs):
        kwargs.setdefault("name", "weights2d")
        super(Weights2D, self).__init__(workflow, **kwargs)
        self.input = None               # weights Array
        self.limit = kwargs.get("limit", 64)
        self.reshape_to = kwargs.get("reshape_to", None)

    def _kernels(self):
        w = self.input
        if hasattr(w, "mem"):
            w.map_read()
            w = w.mem
        w = numpy.asarray(w)
        if self.reshape_to is not None:
            w = w.reshape((-1,) + tuple(self.reshape_to))
        if w.ndim == 2:                  # FC: try square images
            side = int(numpy.sqrt(w.shape[1]))
            if side * side == w.shape[1]:
                w = w.reshape(-1, side, side)
            else:
                w = w[:, None, :]
        if w.ndim == 4:                  # conv [K, ky, kx, C] -> gray
            w = w.mean(axis=-1)
        return w[:self.limit]

    def record(self):
        k = self._kernels()
        return {"kernels": int(k.shape[0]),
                "w_min": float(k.min()), "w_max": float(k.max())}

    def render(self, ax):
        k = self._kernels()
        n = k.shape[0]
        cols = int(numpy.ceil(numpy.sqrt(n)))
        rows = int(numpy.ceil(n / cols))
        kh, kw = k.shape[1], k.shape[2]
        grid = numpy.zeros((rows * (kh + 1), cols * (kw + 1)),
                           numpy.float32)
        for i in range(n):
            r, c = divmod(i, cols)
            grid[r * (kh + 1):r * (kh + 1) + kh,
                 c * (kw + 1):c * (kw + 1) + kw] = k[i]
        ax.imshow(grid, cmap="gray", interpolation="nearest")
        ax.set_title(self.plot_name)
        ax.axis("off")


class MSEHistogram(Histogram):
    """Histogram of per-sample MSE (linked from EvaluatorMSE.mse)."""

    def __init__(self, workflow, **kwargs):
        kwargs.setdefault("name", "mse_hist")
        super(MSEHistogram, self).__init__(workflow, **kwargs)


class KohonenHits(Plotter):
    """Winner-hit counts on the SOM grid."""

    def __init__(self, workflow, **kwargs):
        kwargs.setdefault("name", "kohonen_hits")
        super(KohonenHits, self).__init__(workflow, **kwargs)
        self.input = None               # winners Array (int32)
        self.shape = kwargs.get("shape", (8, 8))
        self._hits = numpy.zeros(self.shape[0] * self.shape[1],
                                 numpy.int64)

    def record(self):
        w = self.input
        if hasattr(w, "mem"):
            w.map_read()
            w = w.mem
        numpy.add.at(self._hits, numpy.asarray(w).reshape(-1), 1)
        return {"total": int(self._hits.sum()),
                "occupied": int((self._hits > 0).sum())}

    def render(self, ax):
        im = ax.imshow(self._hits.reshape(self.shape[::-1]),
                       cmap="hot", interpolation="nearest")
        ax.figure.colorbar(im, ax=ax)
        ax.set_title("SOM hits")


class KohonenNeighborMap(Plotter):
    """U-matrix style neighbor-distance map of SOM weights."""

    def __init__(self, workflow, **kwargs):
        kwargs.setdefault("name", "kohonen_umatrix")
        super(KohonenNeighborMap, self).__init__(workflow, **kwargs)
        self.input = None               # weights Array [n, in]
        self.shape = kwargs.get("shape", (8, 8))

    def _umatrix(self):
        w = self.input
        if hasattr(w, "mem"):
            w.map_read()
            w = w.mem
        sx, sy = self.shape
        grid = numpy.asarray(w).reshape(sy, sx, -1)
        u = numpy.zeros((sy, sx), numpy.float32)
        for y in range(sy):
            for x in range(sx):
                dists = []
                for dy, dx in ((0, 1), (1, 0), (0, -1), (-1, 0)):
                    ny, nx = y + dy, x + dx
                    if 0 <= ny < sy and 0 <= nx < sx:
                        dists.append(numpy.linalg.norm(
                            grid[y, x] - grid[ny, nx]))
                u[y, x] = numpy.mean(dists) if dists else 0
        return u

    def record(self):
        u = self._umatrix()
        return {"u_mean": float(u.mean()), "u_max": float(u.max())}

    def render(self, ax):
        im = ax.imshow(self._umatrix(), cmap="bone",
                       interpolation="nearest")
        ax.figure.colorbar(im, ax=ax)
        ax.set_title("SOM U-matrix")
