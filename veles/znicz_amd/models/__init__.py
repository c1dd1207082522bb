"""Sample model workflows (parity: reference ``znicz/samples/``).

Every sample follows the reference convention: a Workflow subclass plus
``run(load, main)`` driven by ``veles.__main__`` and a ``*_config.py``
mutating ``root.<sample>``.  Datasets are synthetic (deterministic,
seeded) because this environment has no network — BASELINE.json names
synthetic data for every benchmark config.
"""
