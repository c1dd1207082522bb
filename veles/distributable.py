"""Distribution hooks.

Parity target: reference ``veles/distributable.py`` (``Distributable``,
``IDistributable``, ``TriviallyDistributable``) — the per-unit
master/slave serialization contract.

MI355X redesign: the asynchronous ZeroMQ parameter-server plane is
replaced by synchronous in-node data parallelism over RCCL
(``veles.znicz_amd.parallel``).  The hook *names* are preserved so user
units that implemented them still load, and the DP engine maps them:

* ``generate_data_for_master``  -> per-rank partial metrics that Decision
  reduces with one small all-reduce;
* ``apply_data_from_master``    -> no-op (weights are replicated and kept
  in sync by the synchronous gradient all-reduce);
* ``drop_slave_state``          -> kept as a no-op for API parity.
"""


class Distributable(object):
    """Base mixin declaring the distribution protocol hooks."""

    negotiates_on_connect = False

    def generate_data_for_master(self):
        return None

    def generate_data_for_slave(self, slave=None):
        return None

    def apply_data_from_master(self, data):
        pass

    def apply_data_from_slave(self, data, slave=None):
        pass

    def drop_slave_state(self, slave=None):
        pass


class TriviallyDistributable(Distributable):
    """A unit with no distributed state (all hooks are no-ops)."""


class IDistributable(object):
    """Marker kept for zope.interface-free API parity."""
