"""Workflow DAG export.

Parity target: reference ``veles/graph_plotter.py`` /
``Workflow.generate_graph`` — emits Graphviz DOT text (and a PNG when
matplotlib+networkx-free rendering isn't possible, the .dot is the
artifact; no pydot in this environment).
"""


def workflow_to_dot(workflow, name=None):
    name = name or type(workflow).__name__
    lines = ["digraph %s {" % name, "  rankdir=TB;",
             '  node [shape=box, fontsize=10];']
    units = list(workflow)
    ids = {u: "u%d" % i for i, u in enumerate(units)}
    for u in units:
        label = "%s\\n(%s)" % (getattr(u, "name", None) or
                               type(u).__name__, type(u).__name__)
        lines.append('  %s [label="%s"];' % (ids[u], label))
    for u in units:
        for src in getattr(u, "links_from", {}):
            if src in ids:
                lines.append("  %s -> %s;" % (ids[src], ids[u]))
    lines.append("}")
    return "\n".join(lines)


def save_dot(workflow, path):
    with open(path, "w") as f:
        f.write(workflow_to_dot(workflow))
    return path
