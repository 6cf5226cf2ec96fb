"""Transparent capture patches applied inside `AutoDist.scope()`.

Reference behavior: autodist/patch.py:40-116 monkey-patches TF optimizers
(`__init__`/`apply_gradients`) and Keras session machinery so an unmodified
user script is captured. The torch idiom: while a scope is active,

  * `torch.nn.Module.__init__` records every constructed module into the
    default GraphItem; GraphItem.prepare() keeps only top-level ones,
  * `torch.optim.Optimizer.__init__` records every optimizer.

Nothing is patched outside an active scope, and patches are removed on exit.
"""
import threading

import torch

from autodist_amd import graph_item as gi

_lock = threading.Lock()
_depth = 0
_orig_module_init = None
_orig_optimizer_init = None


class PatchTorch:
    """Scope-local patch manager (reference PatchTensorFlow, patch.py:40)."""

    @staticmethod
    def patch():
        global _depth, _orig_module_init, _orig_optimizer_init
        with _lock:
            _depth += 1
            if _depth > 1:
                return
            _orig_module_init = torch.nn.Module.__init__
            _orig_optimizer_init = torch.optim.Optimizer.__init__

            def module_init(self, *args, **kwargs):
                _orig_module_init(self, *args, **kwargs)
                item = gi.get_default_graph_item()
                if item is not None:
                    item.record_module(self)

            def optimizer_init(self, *args, **kwargs):
                _orig_optimizer_init(self, *args, **kwargs)
                item = gi.get_default_graph_item()
                if item is not None:
                    item.record_optimizer(self)

            torch.nn.Module.__init__ = module_init
            torch.optim.Optimizer.__init__ = optimizer_init

    @staticmethod
    def unpatch():
        global _depth
        with _lock:
            _depth -= 1
            if _depth > 0:
                return
            if _orig_module_init is not None:
                torch.nn.Module.__init__ = _orig_module_init
            if _orig_optimizer_init is not None:
                torch.optim.Optimizer.__init__ = _orig_optimizer_init
