"""PS synchronizer — shard-owner parameter-server rounds over xGMI.

Reference behavior: autodist/kernel/synchronization/ps_synchronizer.py (the
reference's biggest kernel, 761 LoC): local dense aggregation (AddN+RealDiv,
460-474), cross-worker ConditionalAccumulator with count-gated take_grad
(556-633), FIFOQueue token barriers for sync (335-385), staleness-bounded
queues (388-458), ProxyVariable caching (537-554).

MI355X-native re-derivation (single xGMI node, one rank per GPU):

  * The owner rank holds the MASTER copy of its shard + shard-local
    optimizer state (replaces PS-device variables + accumulators).
  * One round = reduce(grad -> owner) + owner fused-apply + broadcast(master)
    — all enqueued on the engine's comm HIP stream, so rounds pipeline with
    compute (replaces accumulator/apply/read dataflow).
  * SYNC barrier semantics: with staleness=0 the round's broadcast is
    consumed before the next step's forward (the token-queue barrier,
    335-385). With staleness=k, up to k rounds stay in flight — workers run
    ahead on bounded-stale weights (the staleness queues, 388-458). With
    sync=False rounds are consumed opportunistically up to a queue cap
    (async PS, 261-262).
  * ProxyVariable (537-554): every rank's live parameter IS the local proxy;
    the post-update broadcast into the staging buffer + stream-ordered copy
    is the proxy refresh. local_replication=False keeps the same traffic but
    the refresh is applied lazily at consume time (direct read semantics).
"""
import torch
import torch.distributed as dist

from autodist_amd.parallel import apply as apply_mod

ASYNC_PS_MAX_DEPTH = 4  # queue cap for sync=False (unbounded staleness)


class PSRound:
    """One in-flight reduce->apply->broadcast round."""

    def __init__(self, shard, event=None, handle=None):
        self.shard = shard
        self.event = event     # GPU: comm-stream event after broadcast
        self.handle = handle   # CPU: async work handle


class PSSynchronizer:
    """Between-graph synchronization for PS-kind shards (reference
    between_graph_apply, ps_synchronizer.py:250-332)."""

    @staticmethod
    def issue_round(engine, plan, sh):
        grad = plan.param.grad
        if grad is None:
            return
        gview = sh.slice.view(grad) if sh.slice else grad
        gbuf = gview if gview.is_contiguous() else gview.contiguous()

        def round_body():
            gbuf.mul_(engine.grad_scale())
            dist.reduce(gbuf, dst=sh.owner_rank, group=engine.process_group)
            if engine.rank == sh.owner_rank:
                apply_mod.apply_dense(plan.cls_name, [sh.master], [gbuf],
                                      [sh.state], plan.hyper)
                src_buf = sh.master
            else:
                src_buf = sh.stage
            dist.broadcast(src_buf, src=sh.owner_rank,
                           group=engine.process_group)
            if engine.rank == sh.owner_rank:
                sh.stage.copy_(sh.master)

        key = engine._ps_key(sh)
        if not engine.collectives_active:
            apply_mod.apply_dense(plan.cls_name, [sh.master], [gbuf],
                                  [sh.state], plan.hyper)
            sh.stage.copy_(sh.master)
            engine._ps_outstanding[key].append(PSRound(sh))
            return
        if engine.device.type == "cuda":
            ev = torch.cuda.Event()
            ev.record()
            engine.comm_stream.wait_event(ev)
            with torch.cuda.stream(engine.comm_stream):
                round_body()
            done = torch.cuda.Event()
            done.record(engine.comm_stream)
            engine._ps_outstanding[key].append(PSRound(sh, event=done))
        else:
            # CPU/gloo: async handles so non-owner workers RUN AHEAD within
            # the staleness bound (the c9-verified behavior) — only the owner
            # blocks, on the reduced gradient it must apply.
            gbuf.mul_(engine.grad_scale())
            h_red = dist.reduce(gbuf, dst=sh.owner_rank,
                                group=engine.process_group, async_op=True)
            if engine.rank == sh.owner_rank:
                h_red.wait()
                apply_mod.apply_dense(plan.cls_name, [sh.master], [gbuf],
                                      [sh.state], plan.hyper)
                sh.stage.copy_(sh.master)
                src_buf = sh.master
            else:
                src_buf = sh.stage
            h_bc = dist.broadcast(src_buf, src=sh.owner_rank,
                                  group=engine.process_group, async_op=True)
            engine._ps_outstanding[key].append(
                PSRound(sh, handle=(h_red, h_bc)))

    @staticmethod
    def consume_due_rounds(engine, plan, sh):
        """Pop rounds past the staleness bound; install results into the live
        parameter (reference staleness queues, ps_synchronizer.py:388-458;
        sync token barrier 335-385)."""
        key = engine._ps_key(sh)
        rounds = engine._ps_outstanding[key]
        depth = sh.staleness if sh.sync else ASYNC_PS_MAX_DEPTH
        while len(rounds) > depth:
            r = rounds.pop(0)
            if r.event is not None:
                torch.cuda.current_stream().wait_event(r.event)
            if r.handle is not None:
                for h in (r.handle if isinstance(r.handle, tuple)
                          else (r.handle,)):
                    if h is not None:
                        h.wait()
            view = sh.slice.view(plan.param.data) if sh.slice \
                else plan.param.data
            view.copy_(sh.stage)
