"""CLI entry point.

Parity target: reference ``veles/__main__.py`` + ``scripts/velescli.py``
(``Main``): ``python -m veles <workflow.py> [<config.py> ...] [flags]``
imports the workflow module by path, executes the config files (they
mutate the global ``root``), then invokes the module's
``run(load, main)`` — the sample convention kept from the reference.

MI355X additions: ``--gpus N`` self-spawns one process per GPU via
``torch.distributed.run`` (RCCL inside one node) instead of the
reference's master/slave flags; ``--optimize GENS`` runs the genetics
GA over ``Tune()`` config leaves (the reference's multi-run tuning
flow, in-process).
"""

import argparse
import os
import sys


def parse_args(argv):
    p = argparse.ArgumentParser(
        prog="veles",
        description="MI355X-native Veles/Znicz: unit-graph NN training")
    p.add_argument("workflow", help="workflow .py file (defines "
                                    "run(load, main))")
    p.add_argument("configs", nargs="*", help="config .py files "
                   "(executed; mutate veles.config.root) and/or "
                   "key=value overrides")
    p.add_argument("-b", "--backend", default=None,
                   choices=[None, "rocm", "numpy"],
                   help="compute backend (default: auto)")
    p.add_argument("-s", "--snapshot", default=None,
                   help="resume from this snapshot file")
    p.add_argument("--gpus", type=int, default=0,
                   help="self-spawn N data-parallel ranks (one per GPU)")
    p.add_argument("--dry-run", action="store_true",
                   help="build + initialize the workflow, then exit")
    p.add_argument("--optimize", type=int, default=0, metavar="GENS",
                   help="GA hyper-tuning: run GENS generations over "
                        "Tune()-marked config leaves (each individual "
                        "= one full workflow run)")
    p.add_argument("--population", type=int, default=8,
                   help="GA population size (with --optimize)")
    p.add_argument("--random-seed", type=int, default=None)
    p.add_argument("--dump-graph", default=None, metavar="DOT",
                   help="write the workflow DAG to this .dot file")
    p.add_argument("-v", "--verbose", action="store_true")
    return p.parse_args(argv)


def apply_configs(args):
    from veles.config import root
    for item in args.configs:
        if item.endswith(".py") and os.path.exists(item):
            with open(item) as f:
                code = f.read()
            exec(compile(code, item, "exec"),
                 {"root": root, "__file__": item})
        elif "=" in item:
            path, _, value = item.partition("=")
            node = root
            parts = path.split(".")
            if parts[0] == "root":
                parts = parts[1:]
            for part in parts[:-1]:
                node = getattr(node, part)
            try:
                value = eval(value, {}, {})
            except Exception:
                pass
            setattr(node, parts[-1], value)
        else:
            raise SystemExit("config argument %r is neither a .py file "
                             "nor key=value" % item)


def spawn_ranks(args):
    """Re-exec under torch.distributed.run with one rank per GPU."""
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node", str(args.gpus),
           "--master-addr", "127.0.0.1", "--master-port", "29411",
           "-m", "veles", args.workflow] + args.configs
    if args.backend:
        cmd += ["--backend", args.backend]
    if args.snapshot:
        cmd += ["--snapshot", args.snapshot]
    if args.random_seed is not None:
        cmd += ["--random-seed", str(args.random_seed)]
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    os.execvpe(cmd[0], cmd, os.environ)


class Main(object):
    def __init__(self, args):
        self.args = args
        self.launcher = None
        self.workflow = None

    def load(self, workflow_class, **kwargs):
        from veles.launcher import Launcher
        from veles.snapshotter import SnapshotterToFile
        self.launcher = Launcher(backend=self.args.backend,
                                 snapshot=self.args.snapshot)
        if self.args.snapshot:
            self.workflow = SnapshotterToFile.import_(self.args.snapshot)
            self.workflow.workflow = self.launcher
            # the workflow property setter does not register with the
            # launcher; without this boot() finds launcher.workflow None
            self.launcher.add_ref(self.workflow)
        else:
            self.workflow = workflow_class(self.launcher, **kwargs)
        return self.workflow, bool(self.args.snapshot)

    def main(self, **kwargs):
        if self.workflow is None:
            raise RuntimeError("run(load, main): load() was not called")
        if self.args.dump_graph:
            from veles.graph_plotter import save_dot
            save_dot(self.workflow, self.args.dump_graph)
            print("wrote %s" % self.args.dump_graph)
        if self.args.dry_run:
            dev = self.launcher.make_device()
            self.workflow.initialize(device=dev, **kwargs)
            print("dry run: %d units initialized" % len(self.workflow))
            return
        self.launcher.boot(**kwargs)

    def run_module(self):
        from veles.import_file import import_file_as_module
        mod = import_file_as_module(self.args.workflow)
        # reference order: workflow module import (sets root defaults)
        # FIRST, then config files / overrides on top
        apply_configs(self.args)
        if not hasattr(mod, "run"):
            raise SystemExit("%s does not define run(load, main)" %
                             self.args.workflow)
        if self.args.optimize > 0:
            if int(os.environ.get("WORLD_SIZE", 1)) > 1 or \
                    self.args.gpus > 1:
                raise SystemExit(
                    "--optimize runs single-process (each individual "
                    "is a full workflow run); drop --gpus")
            return self._run_genetics(mod)
        mod.run(self.load, self.main)
        return 0

    def _run_genetics(self, mod):
        """GA hyper-tuning: the reference's multi-run optimization flow
        (``scripts/velescli.py --optimize`` driving veles/genetics/)
        in-process: each individual is one full workflow run with the
        Tune() leaves replaced by its genes; fitness is −best
        validation error count (−train error %% when the loader has no
        validation set)."""
        import logging
        from veles.config import root
        from veles.genetics import (find_tunes, set_config_leaf,
                                    GeneticsOptimizer)
        from veles.loader.base import TRAIN
        tunes = find_tunes(root)
        if not tunes:
            raise SystemExit(
                "--optimize: no Tune() leaves in the config; mark "
                "tunable leaves with veles.genetics.Tune(default, "
                "min, max) in a config file")
        log = logging.getLogger("genetics")
        log.info("tuning %d leaves: %s", len(tunes),
                 ", ".join(p for p, _ in tunes))
        args = self.args

        def evaluate(cfg):
            m = Main(args)             # fresh Launcher + workflow
            mod.run(m.load, m.main)
            dec = getattr(m.workflow, "decision", None)
            if dec is not None and dec.min_validation_n_err is not None:
                fit = -float(dec.min_validation_n_err)
            elif dec is not None:
                fit = -float(dec.epoch_n_err_pt[TRAIN])
            else:
                fit = 0.0
            log.info("evaluated %s -> fitness %.6g",
                     {k: round(v, 6) for k, v in sorted(cfg.items())},
                     fit)
            return fit

        ga = GeneticsOptimizer(tunes, evaluate,
                               population=args.population,
                               generations=args.optimize)
        best = ga.run()
        for (path, _), gene in zip(tunes, best.genes):
            set_config_leaf(path, gene)   # root keeps the winner
            log.info("best %s = %.6g", path, gene)
        log.info("best fitness = %.6g over %d generations",
                 best.fitness, len(ga.history))
        return 0


def main(argv=None):
    args = parse_args(argv if argv is not None else sys.argv[1:])
    # without this the units' self.info() epoch lines go to an
    # unconfigured logger and the CLI runs in total silence
    import logging
    from veles.logger import setup_logging
    setup_logging(logging.DEBUG if args.verbose else logging.INFO)
    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        spawn_ranks(args)                      # does not return
    if args.random_seed is not None:
        from veles import prng
        prng.seed_all(args.random_seed)
    if args.backend:
        from veles.config import root
        root.common.engine.backend = args.backend
    return Main(args).run_module()


if __name__ == "__main__":
    sys.exit(main())
