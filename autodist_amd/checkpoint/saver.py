"""Checkpoint Saver — single-node-compatible checkpoints from distributed
training.

Reference behavior: autodist/checkpoint/saver.py:28-133 — the Saver must be
created BEFORE the distributed session is built (saver.py:63-66), registers
into GraphItem.info.savers, and its save/restore run under the transformed
graph but name-map to original variables so checkpoints are interchangeable
with single-node training (verified by restore-with-vanilla-TF in the
reference's cases/c0.py:125-133; here: restore with vanilla torch).

Format: one torch.save file {"model": ..., "optimizer": ...} where
"optimizer" is a torch.optim-compatible state_dict — a plain torch script
can `optimizer.load_state_dict(ckpt["optimizer"])`. Sharded (partitioned /
PS-owned) optimizer state is reassembled to full tensors on save and
re-sharded on restore (the reference's SaveSliceInfo semantics,
partitioner.py:292-346).
"""
import os

import torch

from autodist_amd import graph_item as gi
from autodist_amd.utils import logging


class Saver:
    def __init__(self, graph_item=None):
        self._graph_item = graph_item or gi.get_default_graph_item()
        if self._graph_item is None:
            from autodist_amd.autodist import get_default_autodist
            ad = get_default_autodist()
            if ad is None:
                raise RuntimeError(
                    "Saver needs an active AutoDist scope or graph_item "
                    "(create it before the distributed session, reference "
                    "saver.py:63-66)")
            self._graph_item = ad.graph_item
        self._graph_item.extend_saver(self)

    # -- helpers -----------------------------------------------------------
    def _engine(self):
        engine = getattr(self._graph_item, "_engine", None)
        if engine is not None:
            return engine
        from autodist_amd.autodist import get_default_autodist
        ad = get_default_autodist()
        return ad.engine if ad is not None else None

    def _optimizer_state_dict(self, engine) -> dict:
        """Build a torch.optim-compatible state_dict from engine state."""
        opt_item = self._graph_item.optimizers[0]
        opt = opt_item.optimizer
        if engine is not None:
            full = engine.consolidated_optimizer_state()
        else:
            full = {}
        state = {}
        idx = 0
        name_of = {id(v.param): n for n, v in
                   self._graph_item.variables.items() if v.param is not None}
        param_groups = []
        for group in opt.param_groups:
            g = {k: v for k, v in group.items() if k != "params"}
            g["params"] = []
            for p in group["params"]:
                g["params"].append(idx)
                name = name_of.get(id(p))
                st = dict(full.get(name, {})) if name else {}
                st = {k: (v.cpu() if isinstance(v, torch.Tensor) else v)
                      for k, v in st.items()}
                if st:
                    state[idx] = st
                idx += 1
            param_groups.append(g)
        return {"state": state, "param_groups": param_groups}

    def _load_optimizer_state_dict(self, engine, sd: dict):
        opt_item = self._graph_item.optimizers[0]
        opt = opt_item.optimizer
        name_of = {id(v.param): n for n, v in
                   self._graph_item.variables.items() if v.param is not None}
        full = {}
        idx = 0
        for group in opt.param_groups:
            for p in group["params"]:
                name = name_of.get(id(p))
                st = sd.get("state", {}).get(idx) or sd.get("state", {}).get(
                    str(idx))
                if name and st:
                    full[name] = {
                        k: (torch.as_tensor(v) if not isinstance(
                            v, torch.Tensor) else v)
                        for k, v in st.items()}
                idx += 1
        if engine is not None:
            engine.load_optimizer_state(full)

    # -- public API --------------------------------------------------------
    def save(self, path, global_step=None) -> str:
        """Write a single-node-compatible checkpoint (chief writes; all ranks
        must call — PS shard gathering is collective)."""
        engine = self._engine()
        if engine is not None:
            engine.drain()
        if global_step is not None:
            path = f"{path}-{global_step}"
        model_sd = {}
        for i, m in enumerate(self._graph_item.models):
            prefix = "" if len(self._graph_item.models) == 1 else f"model{i}."
            for k, v in m.state_dict().items():
                model_sd[prefix + k] = v.cpu()
        opt_sd = self._optimizer_state_dict(engine) \
            if self._graph_item.optimizers else None
        is_chief = engine is None or engine.rank == 0
        if is_chief:
            os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
            torch.save({"model": model_sd, "optimizer": opt_sd,
                        "autodist_version": 1}, path)
            logging.info("checkpoint saved to %s", path)
        return path

    def restore(self, path):
        """Load a (possibly vanilla single-node) checkpoint into the
        distributed state (all ranks call with the same path)."""
        ckpt = torch.load(path, map_location="cpu", weights_only=False)
        model_sd = ckpt.get("model", ckpt)
        engine = self._engine()
        models = self._graph_item.models
        for i, m in enumerate(models):
            prefix = "" if len(models) == 1 else f"model{i}."
            sub = {k[len(prefix):]: v for k, v in model_sd.items()
                   if k.startswith(prefix)}
            m.load_state_dict(sub)
            if engine is not None:
                m.to(engine.device)
        if engine is not None:
            # refresh PS masters/stages with restored values
            for plan in engine.var_plans:
                for sh in plan.shards:
                    if sh.master is not None:
                        view = sh.slice.view(plan.param.data) if sh.slice \
                            else plan.param.data
                        sh.master.copy_(view)
                        sh.stage.copy_(view)
        if ckpt.get("optimizer") and self._graph_item.optimizers:
            self._load_optimizer_state_dict(engine, ckpt["optimizer"])
        logging.info("checkpoint restored from %s", path)
