"""Graph snapshots at each transform phase.

Reference behavior: autodist/utils/visualization_util.py:24-36 writes
TensorBoard graph dumps after each GraphTransformer phase (0-original,
1-after-partition, 2-after-in-graph, 3-transformed). The torch analog dumps
a readable text description of the synchronization plan + model structure
per phase under DEFAULT_GRAPH_DUMP_DIR.
"""
import os
from datetime import datetime

from autodist_amd.const import DEFAULT_GRAPH_DUMP_DIR
from autodist_amd.utils import logging


def log_graph(name: str, content: str) -> str:
    os.makedirs(DEFAULT_GRAPH_DUMP_DIR, exist_ok=True)
    ts = datetime.now().strftime("%Y%m%d-%H%M%S-%f")
    path = os.path.join(DEFAULT_GRAPH_DUMP_DIR, f"{ts}-{name}.txt")
    with open(path, "w", encoding="utf-8") as f:
        f.write(content)
    logging.debug("graph snapshot %s -> %s", name, path)
    return path


def describe_plan(engine) -> str:
    """Readable description of an engine's synchronization plan."""
    lines = [f"world_size={engine.world_size} device={engine.device}"]
    for b in engine.buckets:
        lines.append(
            f"bucket {b.id}: {len(b.params)} vars, {b.numel} elems, "
            f"dtype={b.dtype}, opt={b.cls_name}, "
            f"compressor={type(b.compressor).__name__}")
    for plan in engine.var_plans:
        if plan.bucketed:
            continue
        for sh in plan.shards:
            sl = f"[{sh.slice.start}:{sh.slice.end}]@ax{sh.slice.axis}" \
                if sh.slice else "whole"
            extra = f" owner={sh.owner_rank} sync={sh.sync} " \
                    f"staleness={sh.staleness}" if sh.kind == "ps" else ""
            lines.append(f"{plan.name} {sl}: {sh.kind}"
                         f"{' sparse' if plan.sparse else ''}{extra}")
    return "\n".join(lines)
