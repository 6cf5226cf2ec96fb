"""SavedModel-equivalent export: a self-contained inference artifact.

Reference behavior: autodist/checkpoint/saved_model_builder.py:31-64 — a thin
wrapper over TF's SavedModelBuilder that REQUIRES an AutoDist saver so the
exported variables are the original (master-replica) ones.

Torch translation: exports {dir}/model.pt (consolidated weights via the
AutoDist Saver), {dir}/metadata.json (signatures: input/output specs), and a
TorchScript trace when example inputs are given.
"""
import json
import os

import torch

from autodist_amd.checkpoint.saver import Saver
from autodist_amd.utils import logging


class SavedModelBuilder:
    def __init__(self, export_dir: str):
        self.export_dir = export_dir
        os.makedirs(export_dir, exist_ok=True)

    def add_meta_graph_and_variables(self, saver: Saver, tags=None,
                                     signature_def_map=None,
                                     example_inputs=None):
        """Requires an AutoDist saver (reference saved_model_builder.py:38-44)."""
        if not isinstance(saver, Saver):
            raise ValueError("SavedModelBuilder requires an autodist_amd "
                             "checkpoint Saver")
        self._saver = saver
        self._tags = list(tags or ["serve"])
        self._signatures = signature_def_map or {}
        self._example_inputs = example_inputs
        return self

    def save(self) -> str:
        ckpt_path = os.path.join(self.export_dir, "model.pt")
        self._saver.save(ckpt_path)
        meta = {"tags": self._tags,
                "signatures": {k: str(v) for k, v in self._signatures.items()},
                "format": "autodist_amd.saved_model.v1"}
        gi = self._saver._graph_item
        engine = self._saver._engine()
        is_chief = engine is None or engine.rank == 0
        if is_chief:
            with open(os.path.join(self.export_dir, "metadata.json"),
                      "w", encoding="utf-8") as f:
                json.dump(meta, f, indent=1)
            if self._example_inputs is not None and gi.models:
                model = gi.models[0]
                was_training = model.training
                model.eval()
                try:
                    with torch.no_grad():
                        traced = torch.jit.trace(model, self._example_inputs)
                    traced.save(os.path.join(self.export_dir,
                                             "model_traced.pt"))
                except Exception as exc:  # noqa: BLE001 - trace is best-effort
                    logging.warning("torchscript trace failed: %s", exc)
                finally:
                    model.train(was_training)
            logging.info("saved model exported to %s", self.export_dir)
        return self.export_dir
