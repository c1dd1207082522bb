"""Test fixtures (parity: reference ``veles/dummy.py``)."""

from veles.units import Unit
from veles.workflow import Workflow


class DummyLauncher(object):
    """Stands in for Launcher in unit tests."""

    def __init__(self, **kwargs):
        self.stopped = False
        self.device = None

    def add_ref(self, unit):
        pass

    def del_ref(self, unit):
        pass

    def on_workflow_finished(self):
        self.stopped = True

    def stop(self):
        self.stopped = True

    @property
    def is_slave(self):
        return False

    @property
    def is_master(self):
        return False

    @property
    def is_standalone(self):
        return True


class DummyWorkflow(Workflow):
    """A Workflow with a DummyLauncher parent — unit-test scaffold."""

    def __init__(self, **kwargs):
        super(DummyWorkflow, self).__init__(None, **kwargs)
        self._launcher = DummyLauncher()

    @property
    def launcher(self):
        return self._launcher


class DummyUnit(Unit):
    """Unit whose attributes can be freely assigned in tests."""

    def __init__(self, workflow=None, **kwargs):
        if workflow is None:
            workflow = DummyWorkflow()
        super(DummyUnit, self).__init__(workflow)
        for k, v in kwargs.items():
            setattr(self, k, v)

    def initialize(self, **kwargs):
        super(DummyUnit, self).initialize(**kwargs)

    def run(self):
        pass
