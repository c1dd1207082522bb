"""Interactive shell unit.

Parity target: reference ``veles/interaction.py`` (``Shell`` — embeds
IPython mid-graph for debugging).  Falls back to ``code.interact`` when
IPython is absent (it is not installed in this image); ``enabled=False``
(the default under automation) makes it a no-op passthrough.
"""

from veles.units import Unit


class Shell(Unit):
    def __init__(self, workflow, **kwargs):
        super(Shell, self).__init__(workflow, **kwargs)
        self.enabled = kwargs.get("enabled", False)
        self.banner = kwargs.get("banner",
                                 "veles shell — `wf` is the workflow")

    def run(self):
        if not self.enabled:
            return
        ns = {"wf": self.workflow, "unit": self}
        try:
            from IPython import embed
            embed(user_ns=ns, banner1=self.banner)
        except ImportError:
            import code
            code.interact(banner=self.banner, local=ns)
