"""GraphItem IR — captured model/optimizer metadata.

Re-imagining of the reference's GraphItem (autodist/graph_item.py:218-553).
The reference wraps a tf.Graph plus a grad->target map and variable Info.
PyTorch has no VarHandleOp graph, so the MI355X-native IR captures:

  * every trainable variable: name, shape, dtype, sparse-gradient flag
    (the reference's IndexedSlices distinction, graph_item.py:440-454),
  * the optimizer: class + constructor defaults + per-group params
    (the reference's wrap_optimizer_* interception, graph_item.py:73-109),
  * the modules themselves (for the transformation kernels to rewrite).

Strategy builders consume ONLY the metadata (names/shapes/dtypes/sparsity), so
they are testable without a GPU, exactly like the reference's builders.
"""
import dataclasses
import json
from collections import OrderedDict
from typing import Dict, List, Optional, Tuple

import torch


@dataclasses.dataclass
class VariableItem:
    """Metadata for one trainable variable (reference Info.variables,
    graph_item.py:112-215)."""
    name: str
    shape: Tuple[int, ...]
    dtype: str
    trainable: bool = True
    is_sparse: bool = False    # gradient arrives as IndexedSlices-equivalent
    param: Optional[torch.nn.Parameter] = dataclasses.field(
        default=None, repr=False, compare=False)

    @property
    def numel(self) -> int:
        n = 1
        for s in self.shape:
            n *= s
        return n

    @property
    def bytesize(self) -> int:
        return self.numel * torch.empty((), dtype=getattr(torch, self.dtype)).element_size()

    def to_dict(self):
        return {"name": self.name, "shape": list(self.shape), "dtype": self.dtype,
                "trainable": self.trainable, "is_sparse": self.is_sparse}

    @classmethod
    def from_dict(cls, d):
        return cls(name=d["name"], shape=tuple(d["shape"]), dtype=d["dtype"],
                   trainable=d.get("trainable", True),
                   is_sparse=d.get("is_sparse", False))


@dataclasses.dataclass
class OptimizerItem:
    """Captured optimizer metadata (reference wrap_optimizer_init,
    graph_item.py:73-91)."""
    cls_name: str
    defaults: dict
    param_names: List[str]
    optimizer: Optional[torch.optim.Optimizer] = dataclasses.field(
        default=None, repr=False, compare=False)

    def to_dict(self):
        safe_defaults = {}
        for k, v in self.defaults.items():
            if isinstance(v, (int, float, bool, str, type(None), list, tuple)):
                safe_defaults[k] = v
        return {"cls_name": self.cls_name, "defaults": safe_defaults,
                "param_names": list(self.param_names)}

    @classmethod
    def from_dict(cls, d):
        return cls(cls_name=d["cls_name"], defaults=d.get("defaults", {}),
                   param_names=list(d.get("param_names", [])))


_default_graph_item = None


class GraphItem:
    """The captured training graph metadata (reference GraphItem,
    graph_item.py:218-553)."""

    def __init__(self):
        self._variables: "OrderedDict[str, VariableItem]" = OrderedDict()
        self._optimizers: List[OptimizerItem] = []
        self._models: List[torch.nn.Module] = []
        self._savers: list = []          # checkpoint.Saver registrations
        self._param_to_name: Dict[int, str] = {}
        self._recorded_modules: list = []    # raw scope captures (patch.py)
        self._recorded_optimizers: list = []
        self._prepared = False

    # -- deferred scope capture (reference patch.py interception) ----------
    def record_module(self, module: torch.nn.Module):
        """Record a module constructed under scope(); resolved in prepare()."""
        self._recorded_modules.append(module)

    def record_optimizer(self, optimizer: torch.optim.Optimizer):
        self._recorded_optimizers.append(optimizer)

    # -- capture ----------------------------------------------------------
    def extend_model(self, model: torch.nn.Module, name_prefix: str = ""):
        """Register a model's parameters (called from patch/scope)."""
        if any(m is model for m in self._models):
            return
        self._models.append(model)
        sparse_params = set()
        for mod in model.modules():
            if isinstance(mod, torch.nn.Embedding) and getattr(mod, "sparse", False):
                sparse_params.add(id(mod.weight))
            if isinstance(mod, torch.nn.EmbeddingBag) and getattr(mod, "sparse", False):
                sparse_params.add(id(mod.weight))
        base = f"{name_prefix}." if name_prefix else ""
        for pname, p in model.named_parameters():
            self._register_param(base + pname, p, is_sparse=id(p) in sparse_params)

    def extend_optimizer_info(self, optimizer: torch.optim.Optimizer,
                              defaults: Optional[dict] = None):
        """Register an optimizer (reference extend_optimizer_info,
        graph_item.py:295-299)."""
        for o in self._optimizers:
            if o.optimizer is optimizer:
                return
        names = []
        for group in optimizer.param_groups:
            for p in group["params"]:
                name = self._param_to_name.get(id(p))
                if name is None:
                    name = f"var_{len(self._variables)}"
                    self._register_param(name, p)
                names.append(name)
        self._optimizers.append(OptimizerItem(
            cls_name=type(optimizer).__name__,
            defaults=dict(defaults if defaults is not None else optimizer.defaults),
            param_names=names, optimizer=optimizer))

    def _register_param(self, name: str, p: torch.nn.Parameter, is_sparse=False):
        if id(p) in self._param_to_name:
            return
        if name in self._variables:  # name collision across models
            name = f"{name}_{len(self._variables)}"
        self._param_to_name[id(p)] = name
        self._variables[name] = VariableItem(
            name=name, shape=tuple(p.shape), dtype=str(p.dtype).replace("torch.", ""),
            trainable=p.requires_grad, is_sparse=is_sparse, param=p)

    def mark_sparse(self, name: str, sparse: bool = True):
        self._variables[name].is_sparse = sparse

    def extend_saver(self, saver):
        self._savers.append(saver)

    def prepare(self):
        """Resolve scope captures: register top-level recorded modules (those
        not contained in another recorded module), then optimizers — so every
        optimizer param gets its qualified module name (reference prepare,
        graph_item.py:494-497)."""
        if self._recorded_modules:
            contained = set()
            for m in self._recorded_modules:
                for sub in m.modules():
                    if sub is not m:
                        contained.add(id(sub))
            seen = set()
            for m in self._recorded_modules:
                if id(m) in contained or id(m) in seen:
                    continue
                seen.add(id(m))
                self.extend_model(m)
            self._recorded_modules.clear()
        for opt in self._recorded_optimizers:
            self.extend_optimizer_info(opt)
        self._recorded_optimizers.clear()
        self._prepared = True
        return self

    # -- queries ----------------------------------------------------------
    @property
    def variables(self) -> "OrderedDict[str, VariableItem]":
        return self._variables

    @property
    def trainable_var_op_to_var(self):
        """Name -> VariableItem for trainable vars (reference naming kept for
        builder-code parity, graph_item.py:334-343)."""
        return OrderedDict((k, v) for k, v in self._variables.items() if v.trainable)

    @property
    def optimizers(self) -> List[OptimizerItem]:
        return self._optimizers

    @property
    def optimizer(self) -> Optional[torch.optim.Optimizer]:
        return self._optimizers[0].optimizer if self._optimizers else None

    @property
    def models(self) -> List[torch.nn.Module]:
        return list(self._models)

    @property
    def savers(self):
        return list(self._savers)

    def var_name(self, p: torch.nn.Parameter) -> Optional[str]:
        return self._param_to_name.get(id(p))

    def grad_target_pairs(self):
        """(grad placeholder, target variable) pairs. In torch autograd the
        grad for var v is v.param.grad; we return (name, VariableItem) with
        sparsity flags (reference grad_target_pairs, graph_item.py:440-454)."""
        return [(f"grad/{name}", v) for name, v in
                self.trainable_var_op_to_var.items()]

    # -- scope ------------------------------------------------------------
    def as_default(self):
        return _GraphItemScope(self)

    # -- serialization (metadata only; reference graph_item.py:499-553) ----
    def serialize_to_string(self) -> str:
        return json.dumps({
            "variables": [v.to_dict() for v in self._variables.values()],
            "optimizers": [o.to_dict() for o in self._optimizers],
        }, indent=1)

    @classmethod
    def parse_from_string(cls, s: str) -> "GraphItem":
        d = json.loads(s)
        g = cls()
        for vd in d.get("variables", []):
            v = VariableItem.from_dict(vd)
            g._variables[v.name] = v
        for od in d.get("optimizers", []):
            g._optimizers.append(OptimizerItem.from_dict(od))
        return g


class _GraphItemScope:
    def __init__(self, item: GraphItem):
        self._item = item
        self._prev = None

    def __enter__(self):
        global _default_graph_item
        self._prev = _default_graph_item
        _default_graph_item = self._item
        return self._item

    def __exit__(self, *exc):
        global _default_graph_item
        _default_graph_item = self._prev
        return False


def get_default_graph_item() -> Optional[GraphItem]:
    return _default_graph_item
