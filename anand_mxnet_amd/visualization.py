"""Network visualization (parity with python/mxnet/visualization.py —
print_summary; plot_network emits graphviz dot text)."""
from __future__ import annotations

import numpy as np

__all__ = ["print_summary", "plot_network"]


def print_summary(symbol, shape=None):
    """Print a layer table for a Symbol (reference print_summary)."""
    shapes = {}
    if shape:
        arg_shapes, _, _ = symbol.infer_shape(**shape)
        shapes = dict(zip(symbol.list_arguments(), arg_shapes))
    print(f"{'Layer (type)':<40} {'Output Shape':<20} {'Param #':<10}")
    print("=" * 72)
    total = 0
    for node in symbol._topo():
        if node.op == "null":
            continue
        nparams = 0
        for inp in node.inputs:
            if inp.op == "null" and inp.name in shapes and \
                    shapes[inp.name] is not None and \
                    not inp.name.endswith(("data", "label")):
                nparams += int(np.prod(shapes[inp.name]))
        total += nparams
        print(f"{node.name + ' (' + node.op + ')':<40} "
              f"{'':<20} {nparams:<10}")
    print("=" * 72)
    print(f"Total params: {total}")
    return total


def print_block_summary(block, input_shape=None):
    """Parameter table for a Gluon block."""
    print(f"{'Parameter':<50} {'Shape':<22} {'#':<12}")
    print("=" * 86)
    total = 0
    for name, p in block.collect_params().items():
        n = int(np.prod(p.shape)) if p.shape else 0
        total += n
        print(f"{name:<50} {str(p.shape):<22} {n:<12}")
    print("=" * 86)
    print(f"Total params: {total}")
    return total


def plot_network(symbol, title="plot", shape=None):
    """Emit a graphviz dot description (no graphviz dependency needed to
    just write the file)."""
    lines = [f'digraph "{title}" {{', "  rankdir=BT;"]
    order = symbol._topo()
    idx = {id(n): i for i, n in enumerate(order)}
    for n in order:
        shape_attr = "ellipse" if n.op == "null" else "box"
        lines.append(f'  n{idx[id(n)]} [label="{n.name}\\n{n.op}" '
                     f'shape={shape_attr}];')
        for inp in n.inputs:
            lines.append(f"  n{idx[id(inp)]} -> n{idx[id(n)]};")
    lines.append("}")
    return "\n".join(lines)
