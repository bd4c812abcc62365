"""Symbolic graph API (parity role of mxnet.sym + nnvm Symbol).

MI355X-native design: the reference carries a full nnvm graph compiler
(3rdparty/tvm/nnvm) because its executor plans memory and fuses engine ops
ahead of time.  Here execution is stream-ordered eager + hipGraph step
capture, so the Symbol's only remaining jobs are (1) a declarative graph
description for the Module API, (2) the ``-symbol.json`` checkpoint format
(save/load round-trip), and (3) shape/dtype inference so Module.bind can
allocate parameters.  The JSON schema mirrors the reference
(nodes/arg_nodes/heads with [id, out_idx, version] links).

Ops cover the classic Module-era set; any registered nd op can be used
symbolically (executed via ops.registry at eval time), shape inference
covers the common layers and composes elementwise by default.
"""
from __future__ import annotations

import ast
import json

import numpy as np

from .base import MXNetError
from . import ndarray as nd
from .ndarray.ndarray import NDArray

__all__ = ["Symbol", "var", "Variable", "load", "loads", "Group",
           "trace_block"]


def _attr_str(v):
    return str(v)


def _parse_attr(s):
    if not isinstance(s, str):
        return s
    try:
        return ast.literal_eval(s)
    except (ValueError, SyntaxError):
        return s


class Symbol:
    """A node in the symbolic graph (op output or variable)."""

    def __init__(self, op, name, inputs, attrs=None):
        self.op = op          # op name, or 'null' for variables
        self.name = name
        self.inputs = inputs  # list of Symbol
        self.attrs = attrs or {}

    # ---------------- graph construction ----------------
    def __repr__(self):
        return f"<Symbol {self.name}>"

    def __add__(self, other):
        return _make_op("elemwise_add", [self, _as_sym(other)])

    def __sub__(self, other):
        return _make_op("elemwise_sub", [self, _as_sym(other)])

    def __mul__(self, other):
        return _make_op("elemwise_mul", [self, _as_sym(other)])

    def __truediv__(self, other):
        return _make_op("elemwise_div", [self, _as_sym(other)])

    def get_internals(self):
        out = []
        for node in self._topo():
            out.append(node)
        return _SymbolList(out)

    def list_arguments(self):
        return [n.name for n in self._topo() if n.op == "null"
                and not n.attrs.get("__aux__")]

    def list_auxiliary_states(self):
        return [n.name for n in self._topo() if n.op == "null"
                and n.attrs.get("__aux__")]

    def list_outputs(self):
        return [self.name + "_output"]

    def _topo(self):
        seen, order = set(), []

        def visit(node):
            if id(node) in seen:
                return
            seen.add(id(node))
            for i in node.inputs:
                visit(i)
            order.append(node)
        visit(self)
        return order

    # ---------------- shape inference ----------------
    def infer_shape(self, **kwargs):
        shapes = {}
        for node in self._topo():
            if node.op == "null":
                if node.name in kwargs:
                    shapes[id(node)] = tuple(kwargs[node.name])
                elif "__shape__" in node.attrs:
                    shapes[id(node)] = tuple(_parse_attr(
                        node.attrs["__shape__"]))
                else:
                    shapes[id(node)] = None
            else:
                in_shapes = [shapes[id(i)] for i in node.inputs]
                out, fixed_ins = _infer_shape_op(node, in_shapes)
                shapes[id(node)] = out
                for inp, s in zip(node.inputs, fixed_ins):
                    if shapes[id(inp)] is None and s is not None:
                        shapes[id(inp)] = tuple(s)
        arg_shapes = [shapes[id(n)] for n in self._topo() if n.op == "null"
                      and not n.attrs.get("__aux__")]
        out_shapes = [shapes[id(self)]]
        aux_shapes = [shapes[id(n)] for n in self._topo() if n.op == "null"
                      and n.attrs.get("__aux__")]
        return arg_shapes, out_shapes, aux_shapes

    # ---------------- execution ----------------
    def eval(self, inputs, params=None):
        """Topologically execute with nd ops.  inputs/params: name->NDArray."""
        from .ops import OPS
        params = params or {}
        vals = {}
        for node in self._topo():
            if node.op == "null":
                if node.name in inputs:
                    vals[id(node)] = inputs[node.name]
                elif node.name in params:
                    v = params[node.name]
                    vals[id(node)] = v if isinstance(v, NDArray) else NDArray(v)
                else:
                    raise MXNetError(f"unbound variable {node.name}")
            else:
                args = [vals[id(i)] for i in node.inputs]
                kwargs = {k: _parse_attr(v) for k, v in node.attrs.items()
                          if not k.startswith("__")}
                vals[id(node)] = OPS[node.op](*args, **kwargs)
        return vals[id(self)]

    def simple_bind(self, ctx=None, grad_req="write", **shapes):
        from .executor import Executor
        return Executor(self, ctx, grad_req=grad_req, shapes=shapes)

    def bind(self, ctx, args, args_grad=None, grad_req="write",
             aux_states=None):
        from .executor import Executor
        return Executor(self, ctx, args=args, args_grad=args_grad,
                        grad_req=grad_req, aux_states=aux_states)

    # ---------------- serialization ----------------
    def tojson(self):
        order = self._topo()
        idx = {id(n): i for i, n in enumerate(order)}
        nodes = []
        for n in order:
            nodes.append({
                "op": n.op,
                "name": n.name,
                "attrs": {k: _attr_str(v) for k, v in n.attrs.items()},
                "inputs": [[idx[id(i)], 0, 0] for i in n.inputs],
            })
        arg_nodes = [i for i, n in enumerate(order) if n.op == "null"]
        return json.dumps({
            "nodes": nodes,
            "arg_nodes": arg_nodes,
            "node_row_ptr": list(range(len(order) + 1)),
            "heads": [[idx[id(self)], 0, 0]],
            "attrs": {"mxnet_version": ["int", 10700]},
        }, indent=2)

    def save(self, fname):
        with open(fname, "w") as f:
            f.write(self.tojson())


class _SymbolList(list):
    def __getitem__(self, key):
        if isinstance(key, str):
            for s in self:
                if s.name == key or s.name + "_output" == key:
                    return s
            raise KeyError(key)
        return list.__getitem__(self, key)


def _as_sym(x):
    if isinstance(x, Symbol):
        return x
    raise TypeError(f"cannot compose Symbol with {type(x)}")



def _make_op(op, inputs, attrs=None, name=None):
    from . import attribute, name as _nm
    name = _nm.current().get(name, op.lower())
    return Symbol(op, name, inputs, attribute.current().get(attrs))


def var(name, shape=None, dtype=None, **kwargs):
    attrs = {}
    if shape is not None:
        attrs["__shape__"] = _attr_str(tuple(shape))
    if dtype is not None:
        attrs["__dtype__"] = _attr_str(dtype)
    attrs.update(kwargs)
    return Symbol("null", name, [], attrs)


Variable = var


def Group(symbols):
    return _make_op("_group", list(symbols))


def load(fname):
    with open(fname) as f:
        return loads(f.read())


def loads(json_str):
    data = json.loads(json_str)
    nodes = []
    for nd_ in data["nodes"]:
        attrs = nd_.get("attrs", nd_.get("param", {}))
        sym_node = Symbol(nd_["op"], nd_["name"], [], dict(attrs))
        nodes.append(sym_node)
    for sym_node, nd_ in zip(nodes, data["nodes"]):
        sym_node.inputs = [nodes[i[0]] for i in nd_["inputs"]]
    head = data["heads"][0][0]
    return nodes[head]


# ---------------- op functions on symbols ----------------

# parameter slots auto-created as variables when not passed explicitly
# (reference UX: sym.FullyConnected(x, num_hidden=8, name='fc1') creates
# fc1_weight / fc1_bias automatically)
_AUTO_PARAMS = {
    "FullyConnected": ("weight", "bias"),
    "Convolution": ("weight", "bias"),
    "Deconvolution": ("weight", "bias"),
    "Embedding": ("weight",),
    "BatchNorm": ("gamma", "beta", "moving_mean", "moving_var"),
    "LayerNorm": ("gamma", "beta"),
}


def _sym_op_factory(op_name):
    def fn(*args, name=None, **kwargs):
        inputs = [a for a in args if isinstance(a, Symbol)]
        # keyword-NDArray-style: data=..., weight=...
        explicit = {}
        for key in ("data", "weight", "bias", "gamma", "beta", "moving_mean",
                    "moving_var", "label"):
            if key in kwargs and isinstance(kwargs[key], Symbol):
                explicit[key] = kwargs.pop(key)
        auto = _AUTO_PARAMS.get(op_name, ())
        for key in ("data", "label"):  # non-parameter inputs keep their slot
            if key in explicit:
                inputs.append(explicit.pop(key))
        if auto:
            # resolve the node name now so auto vars get '{name}_{slot}'
            from . import name as _nm
            name = _nm.current().get(name, op_name.lower())
            no_bias = str(kwargs.get("no_bias", False)).lower() in \
                ("true", "1")
            for slot in auto:
                if slot == "bias" and no_bias:
                    continue
                s = explicit.pop(slot, None)
                if s is None:
                    s = var(f"{name}_{slot}")
                inputs.append(s)
        inputs.extend(explicit.values())
        attrs = {k: _attr_str(v) for k, v in kwargs.items()}
        return _make_op(op_name, inputs, attrs, name)
    fn.__name__ = op_name
    return fn


def _install_ops():
    import sys
    from .ops import OPS
    mod = sys.modules[__name__]
    for op_name in OPS:
        if not hasattr(mod, op_name):
            setattr(mod, op_name, _sym_op_factory(op_name))


_install_ops()


# ---------------- shape inference rules ----------------

def _pairi(v):
    v = _parse_attr(v) if isinstance(v, str) else v
    if isinstance(v, int):
        return (v, v)
    return tuple(v)


def _infer_shape_op(node, ins):
    """Returns (out_shape, fixed_input_shapes) — fills unknown param shapes."""
    a = {k: _parse_attr(v) for k, v in node.attrs.items()}
    op = node.op
    fixed = [None] * len(ins)
    x = ins[0] if ins else None
    if op == "FullyConnected":
        num_hidden = int(a["num_hidden"])
        flatten = a.get("flatten", True)
        in_units = int(np.prod(x[1:])) if flatten else x[-1]
        fixed = [None, (num_hidden, in_units)] + \
            ([(num_hidden,)] if len(ins) > 2 else [])
        out = (x[0], num_hidden) if flatten else tuple(x[:-1]) + (num_hidden,)
        return out, fixed
    if op == "Convolution":
        kernel = _pairi(a["kernel"])
        stride = _pairi(a.get("stride", (1, 1)))
        pad = _pairi(a.get("pad", (0, 0)))
        dilate = _pairi(a.get("dilate", (1, 1)))
        nf = int(a["num_filter"])
        groups = int(a.get("num_group", 1))
        layout = a.get("layout", "NCHW")
        if layout == "NCHW":
            N, C, H, W = x
        else:
            N, H, W, C = x
        OH = (H + 2 * pad[0] - dilate[0] * (kernel[0] - 1) - 1) // stride[0] + 1
        OW = (W + 2 * pad[1] - dilate[1] * (kernel[1] - 1) - 1) // stride[1] + 1
        wshape = (nf, C // groups, kernel[0], kernel[1]) if layout == "NCHW" \
            else (nf, kernel[0], kernel[1], C // groups)
        fixed = [None, wshape] + ([(nf,)] if len(ins) > 2 else [])
        out = (N, nf, OH, OW) if layout == "NCHW" else (N, OH, OW, nf)
        return out, fixed
    if op == "Pooling":
        layout = a.get("layout", "NCHW")
        if a.get("global_pool", False):
            if layout == "NCHW":
                return (x[0], x[1], 1, 1), fixed
            return (x[0], 1, 1, x[3]), fixed
        kernel = _pairi(a.get("kernel", (2, 2)))
        stride = _pairi(a.get("stride", kernel))
        pad = _pairi(a.get("pad", (0, 0)))
        if layout == "NCHW":
            N, C, H, W = x
        else:
            N, H, W, C = x
        OH = (H + 2 * pad[0] - kernel[0]) // stride[0] + 1
        OW = (W + 2 * pad[1] - kernel[1]) // stride[1] + 1
        return ((N, C, OH, OW) if layout == "NCHW" else (N, OH, OW, C)), fixed
    if op == "BatchNorm":
        axis = int(a.get("axis", 1))
        ch = x[axis]
        fixed = [None] + [(ch,)] * (len(ins) - 1)
        return tuple(x), fixed
    if op == "LayerNorm":
        ch = x[int(a.get("axis", -1)) % len(x)]
        fixed = [None] + [(ch,)] * (len(ins) - 1)
        return tuple(x), fixed
    if op == "Embedding":
        return tuple(x) + (int(a["output_dim"]),), \
            [None, (int(a["input_dim"]), int(a["output_dim"]))]
    if op == "Flatten":
        return (x[0], int(np.prod(x[1:]))), fixed
    if op == "reshape":
        shape = a.get("shape")
        out = []
        for i, s in enumerate(shape):
            out.append(x[i] if s == 0 else s)
        if -1 in out:
            known = int(np.prod([s for s in out if s != -1]))
            out[out.index(-1)] = int(np.prod(x)) // known
        return tuple(out), fixed
    if op in ("Activation", "relu", "sigmoid", "tanh", "softmax",
              "log_softmax", "Dropout", "LeakyReLU", "SoftmaxOutput",
              "identity", "BlockGrad", "Cast", "negative", "exp", "log",
              "sqrt", "square", "abs"):
        if op == "SoftmaxOutput":
            fixed = [None, (x[0],)]
        return tuple(x), fixed
    if op in ("elemwise_add", "elemwise_sub", "elemwise_mul", "elemwise_div",
              "broadcast_add", "broadcast_sub", "broadcast_mul",
              "broadcast_div", "add_n"):
        return tuple(x), fixed
    if op == "concat":
        dim = int(a.get("dim", 1))
        out = list(x)
        out[dim] = sum(s[dim] for s in ins)
        return tuple(out), fixed
    if op == "transpose":
        axes = a.get("axes") or tuple(reversed(range(len(x))))
        return tuple(x[i] for i in axes), fixed
    if op == "_group":
        return tuple(x), fixed
    # default: shape-preserving
    return (tuple(x) if x is not None else None), fixed


def trace_block(block):
    """Build a Symbol for a HybridBlock by structural walk (export path).

    Covers Sequential-style compositions; blocks with a custom forward are
    exported as a single '_block_call' node (executable within this
    framework via SymbolBlock; cross-framework JSON parity covers the
    classic layer set)."""
    data = var("data")
    return _trace(block, data)


def _trace(block, x):
    from .gluon import nn as gnn
    if isinstance(block, (gnn.HybridSequential, gnn.Sequential)):
        for child in block._children.values():
            x = _trace(child, x)
        return x
    if isinstance(block, gnn.Dense):
        w = var(block.weight.name, shape=block.weight.shape)
        args = [x, w]
        if block.bias is not None:
            args.append(var(block.bias.name, shape=block.bias.shape))
        s = _make_op("FullyConnected", args,
                     {"num_hidden": _attr_str(block._units),
                      "no_bias": _attr_str(block.bias is None),
                      "flatten": _attr_str(block._flatten)},
                     name=block.name + "_fwd")
        if block.act is not None:
            s = _make_op("Activation", [s],
                         {"act_type": _attr_str(block.act._act_type)})
        return s
    if isinstance(block, gnn.Conv2D):
        w = var(block.weight.name, shape=block.weight.shape)
        args = [x, w]
        if block.bias is not None:
            args.append(var(block.bias.name, shape=block.bias.shape))
        s = _make_op("Convolution", args, {
            "kernel": _attr_str(block._kernel),
            "stride": _attr_str(block._strides),
            "pad": _attr_str(block._padding),
            "dilate": _attr_str(block._dilation),
            "num_filter": _attr_str(block._channels),
            "num_group": _attr_str(block._groups),
            "no_bias": _attr_str(block.bias is None),
            "layout": _attr_str(block._layout)},
            name=block.name + "_fwd")
        if block.act is not None:
            s = _make_op("Activation", [s],
                         {"act_type": _attr_str(block.act._act_type)})
        return s
    if isinstance(block, gnn.BatchNorm):
        gamma = var(block.gamma.name, shape=block.gamma.shape)
        beta = var(block.beta.name, shape=block.beta.shape)
        mm = var(block.running_mean.name, shape=block.running_mean.shape,
                 __aux__=True)
        mv = var(block.running_var.name, shape=block.running_var.shape,
                 __aux__=True)
        return _make_op("BatchNorm", [x, gamma, beta, mm, mv],
                        {k: _attr_str(v) for k, v in block._kwargs.items()},
                        name=block.name + "_fwd")
    if isinstance(block, gnn.Activation):
        return _make_op("Activation", [x],
                        {"act_type": _attr_str(block._act_type)})
    if isinstance(block, gnn.Flatten):
        return _make_op("Flatten", [x])
    if isinstance(block, gnn.Dropout):
        return _make_op("Dropout", [x], {"p": _attr_str(block._rate)})
    if isinstance(block, (gnn.MaxPool2D, gnn.AvgPool2D, gnn.GlobalAvgPool2D,
                          gnn.GlobalMaxPool2D)):
        kw = getattr(block, "_kwargs", None)
        if kw is None:
            pt = "avg" if isinstance(block, gnn.GlobalAvgPool2D) else "max"
            kw = {"global_pool": True, "pool_type": pt, "kernel": (1, 1),
                  "layout": block._layout}
        return _make_op("Pooling", [x],
                        {k: _attr_str(v) for k, v in kw.items()})
    raise MXNetError(
        f"export: cannot symbolically trace block {type(block).__name__}; "
        "use save_parameters/load_parameters for this model")
