"""Genetic hyper-parameter optimization.

Parity target: reference ``veles/genetics/`` (``Range``/``Tune`` config
wrappers + the GA core): ``Tune()``-marked config leaves become
chromosome genes; each individual's fitness is the best validation
metric of a full workflow run.

MI355X redesign: evaluations run in-process and sequentially (or via a
user-supplied parallel map) instead of the reference's master/slave
fan-out — the per-GPU process model belongs to the training run itself.
"""

import numpy

from veles.config import Config, root  # noqa: F401


class Range(object):
    """A [min, max] range marker (reference veles/genetics/config.py)."""

    def __init__(self, default, vmin, vmax):
        self.default = default
        self.vmin = vmin
        self.vmax = vmax

    def __repr__(self):
        return "Range(%s, %s, %s)" % (self.default, self.vmin, self.vmax)


class Tune(Range):
    """Marks a config leaf as tunable by the GA."""


def find_tunes(cfg, prefix="root"):
    """Walk a Config tree; return [(path, Tune)] for every Tune leaf."""
    out = []
    for key, val in cfg:
        path = "%s.%s" % (prefix, key)
        if isinstance(val, Tune):
            out.append((path, val))
        elif isinstance(val, Config):
            out.extend(find_tunes(val, path))
    return out


def set_config_leaf(path, value, base=None):
    node = base if base is not None else root
    parts = path.split(".")
    if parts[0] == "root":
        parts = parts[1:]
    for p in parts[:-1]:
        node = getattr(node, p)
    setattr(node, parts[-1], value)


class Chromosome(object):
    def __init__(self, genes):
        self.genes = list(genes)
        self.fitness = None

    def mutate(self, tunes, rng, rate=0.2):
        for i, (_, t) in enumerate(tunes):
            if rng.uniform() < rate:
                sigma = (t.vmax - t.vmin) * 0.1
                g = self.genes[i] + rng.normal(0.0, sigma)
                self.genes[i] = float(numpy.clip(g, t.vmin, t.vmax))

    @classmethod
    def crossover(cls, a, b, rng):
        mask = rng.uniform(size=len(a.genes)) < 0.5
        genes = [ga if m else gb
                 for ga, gb, m in zip(a.genes, b.genes, mask)]
        return cls(genes)


class GeneticsOptimizer(object):
    """GA over Tune leaves.

    evaluate(config_dict) -> fitness (higher is better) is supplied by
    the caller; it typically sets the leaves, builds the workflow, runs
    it for a few epochs and returns −validation_error.
    """

    def __init__(self, tunes, evaluate, population=8, generations=5,
                 elite=2, seed=0xBEEF, apply_leaves=True, pmap=None):
        self.tunes = list(tunes)
        self.evaluate = evaluate
        self.population_size = population
        self.generations = generations
        self.elite = elite
        self.rng = numpy.random.RandomState(seed)
        self.apply_leaves = apply_leaves
        # pmap(func, [cfg, ...]) -> [fitness, ...]: evaluates a whole
        # generation concurrently (e.g. multiprocessing.Pool.map with
        # a module-level evaluate).  With pmap the leaves are NOT
        # applied to the global root here — each worker must apply its
        # own cfg dict (mutating shared root would race).
        self.pmap = pmap
        self.best = None
        self.history = []

    def _random_chromosome(self):
        genes = [float(self.rng.uniform(t.vmin, t.vmax))
                 for _, t in self.tunes]
        return Chromosome(genes)

    def _cfg(self, chrom):
        return {path: g for (path, _), g in zip(self.tunes, chrom.genes)}

    def _eval(self, chrom):
        cfg = self._cfg(chrom)
        if self.apply_leaves:
            for path, val in cfg.items():
                set_config_leaf(path, val)
        chrom.fitness = float(self.evaluate(cfg))
        return chrom.fitness

    def _eval_generation(self, pop):
        todo = [c for c in pop if c.fitness is None]
        if self.pmap is not None and len(todo) > 1:
            fits = list(self.pmap(self.evaluate,
                                  [self._cfg(c) for c in todo]))
            for c, f in zip(todo, fits):
                c.fitness = float(f)
        else:
            for c in todo:
                self._eval(c)

    def run(self):
        pop = [self._random_chromosome()
               for _ in range(self.population_size)]
        for gen in range(self.generations):
            self._eval_generation(pop)
            pop.sort(key=lambda c: c.fitness, reverse=True)
            if self.best is None or pop[0].fitness > self.best.fitness:
                self.best = Chromosome(pop[0].genes)
                self.best.fitness = pop[0].fitness
            self.history.append(pop[0].fitness)
            if gen == self.generations - 1:
                break
            nxt = pop[:self.elite]
            while len(nxt) < self.population_size:
                # parents from the better half; max(2,·) keeps
                # choice(n, 2, replace=False) legal for population < 4
                a, b = self.rng.choice(max(2, len(pop) // 2), 2,
                                       replace=False)
                child = Chromosome.crossover(pop[a], pop[b], self.rng)
                child.mutate(self.tunes, self.rng)
                nxt.append(child)
            pop = nxt
        return self.best
