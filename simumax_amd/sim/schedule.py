"""PpSchedule: build per-rank simulator job lists from the analytic chunks.

Parity target: simumax/core/transformer/pipeline_schedule.py:30-959
(PpSchedule.prefill_batch 1F1B, OptimizerSimulator) — jobs are derived
from the SAME leaf modules/CommEvents the analytic coster priced, so
perf() and simulate() agree by construction. Sync p2p semantics (the
async batched-bundle VPP path is a later extension).
"""

from __future__ import annotations

from typing import Dict, List

from ..core.utils import get_pp_p2p_comm_size, get_pp_stage_representative_rank
from .events import Job, MemDelta


def _leaf_jobs_fwd(chunk, stage, mb, p2p=None):
    jobs = []
    segments = chunk._segments(chunk.leaf_modules())
    for seg in segments:
        for leaf in seg["leaves"]:
            ci = leaf.get_cost_info()
            ai = leaf.get_act_info()
            cache = (leaf.input_info.total_bytes()
                     if seg["recompute"] and leaf is seg["leaves"][0]
                     else (0.0 if seg["recompute"] else ai.activation_mem_cache))
            jobs.append(Job(
                name=leaf.full_name, kind="fwd", dur=ci.fwd_compute_time,
                mb=mb,
                mem=MemDelta(alloc_bytes=cache,
                             transient_bytes=ai.fwd_peak_mem_no_cache,
                             token_key=f"mb{mb}.{leaf.full_name}"),
                call_stack=leaf.full_name,
            ))
            for ev in leaf.comm_ops:
                if ev.stage != "fwd":
                    continue
                jobs.append(Job(
                    name=f"{leaf.full_name}.{ev.op_name}", kind="comm",
                    dur=ev.time_ms, lane="comm", mb=mb,
                    gid=f"mb{mb}-{leaf.full_name}-{ev.op_name}-{ev.comm_stage}",
                    call_stack=leaf.full_name, overlap=ev.overlap,
                ))
    return jobs


def _leaf_jobs_bwd(chunk, stage, mb):
    jobs = []
    segments = chunk._segments(chunk.leaf_modules())
    for seg in reversed(segments):
        if seg["recompute"]:
            # re-forward the segment (RecomputeBlockJob analog)
            for leaf in seg["leaves"]:
                ci = leaf.get_cost_info()
                ai = leaf.get_act_info()
                jobs.append(Job(
                    name=f"{leaf.full_name}(recompute)", kind="recompute",
                    dur=ci.recompute_compute_time, mb=mb,
                    mem=MemDelta(alloc_bytes=ai.activation_mem_cache,
                                 token_key=f"mb{mb}.rc.{leaf.full_name}"),
                    call_stack=leaf.full_name,
                ))
        for leaf in reversed(seg["leaves"]):
            ci = leaf.get_cost_info()
            ai = leaf.get_act_info()
            if seg["recompute"]:
                free = ai.activation_mem_cache
                key = f"mb{mb}.rc.{leaf.full_name}"
            else:
                free = ai.activation_mem_cache
                key = f"mb{mb}.{leaf.full_name}"
            jobs.append(Job(
                name=f"{leaf.full_name}(bwd)", kind="bwd",
                dur=ci.bwd_grad_act_time + ci.bwd_grad_w_time, mb=mb,
                mem=MemDelta(free_bytes=free,
                             transient_bytes=ai.bwd_peak_mem_no_cache,
                             token_key=key),
                call_stack=leaf.full_name,
            ))
            if seg["recompute"] and leaf is seg["leaves"][0]:
                # release the segment-input tensor held since forward
                jobs.append(Job(
                    name=f"{leaf.full_name}(free segment input)", kind="bwd",
                    dur=0.0, mb=mb,
                    mem=MemDelta(free_bytes=leaf.input_info.total_bytes(),
                                 token_key=f"mb{mb}.{leaf.full_name}"),
                    call_stack=leaf.full_name,
                ))
            for ev in leaf.comm_ops:
                if ev.stage not in ("bwd_act", "bwd_w"):
                    continue
                jobs.append(Job(
                    name=f"{leaf.full_name}.{ev.op_name}(bwd)", kind="comm",
                    dur=ev.time_ms, lane="comm", mb=mb,
                    gid=f"mb{mb}-{leaf.full_name}-{ev.op_name}-{ev.stage}",
                    call_stack=leaf.full_name, overlap=ev.overlap,
                ))
    return jobs


def _assemble_units(units, r, async_p2p):
    """Flatten per-unit job lists into one rank stream.

    Sync mode: recv jobs block at their original position (kind "p2p").
    Async mode (strategy.pp_comm_async, Megatron irecv/isend semantics):
    each unit's recv is POSTED one unit ahead (free) and a WAIT sits at
    the original position, so the transfer overlaps the preceding
    compute — reference parity: pipeline_schedule.py:200-267 (batched
    async bundles).
    """
    def is_recv(j):
        return j.kind == "p2p" and j.peers and j.peers[1] == r

    if not async_p2p:
        return [j for u in units for j in u]

    def post_of(j):
        return Job(name=j.name.replace("recv", "post_recv"),
                   kind="p2p_post_recv", dur=0.0, lane="comm", mb=j.mb,
                   gid=j.gid, peers=j.peers, call_stack=j.call_stack)

    def wait_of(j):
        return Job(name=j.name.replace("recv", "wait_recv"),
                   kind="p2p_wait", dur=j.dur, lane="comm", mb=j.mb,
                   gid=j.gid, peers=j.peers, mem=j.mem,
                   call_stack=j.call_stack)

    out = []
    for i, u in enumerate(units):
        if i == 0 and u and is_recv(u[0]):
            out.append(post_of(u[0]))
        nxt = units[i + 1] if i + 1 < len(units) else None
        if nxt and nxt and is_recv(nxt[0]):
            out.append(post_of(nxt[0]))
        for j in u:
            out.append(wait_of(j) if is_recv(j) else j)
    return out


class PpSchedule:
    """Builds the per-(simulated-)rank 1F1B job lists."""

    def __init__(self, perf_model, merge_lanes=True):
        self.perf = perf_model
        self.strategy = perf_model.strategy
        self.system = perf_model.system
        self.merge_lanes = merge_lanes

    def sim_ranks(self) -> List[int]:
        s = self.strategy
        if self.merge_lanes:
            return [get_pp_stage_representative_rank(i, s)
                    for i in range(s.pp_size)]
        return list(range(s.world_size))

    def build(self) -> Dict[int, List[Job]]:
        s = self.perf.strategy
        if max(1, s.interleaving_size) > 1:
            return self.build_interleaved()
        pp, mbc = s.pp_size, s.micro_batch_num
        p2p_size = get_pp_p2p_comm_size(s, self.perf.model_config)
        p2p_time = 0.0
        if pp > 1:
            p2p_time = self.system.compute_net_op_time(
                "p2p", p2p_size, 2, net=s.pp_net, comm_stage="pp",
                strategy=s)

        ranks = self.sim_ranks()
        jobs: Dict[int, List[Job]] = {r: [] for r in ranks}
        per_stage_rank = {i: get_pp_stage_representative_rank(i, s)
                          for i in range(pp)}

        # per-rank 1F1B streams (same order as the analytic recurrence)
        for stage in range(pp):
            rep = per_stage_rank[stage]
            stage_ranks = [r for r in ranks
                           if r // (s.world_size // pp) == stage] or [rep]
            chunk = self.perf.chunks[stage]
            warm = min(pp - stage - 1, mbc)
            stream = [("F", m) for m in range(warm)]
            nf, nb = warm, 0
            while nb < mbc:
                if nf < mbc:
                    stream.append(("F", nf)); nf += 1
                stream.append(("B", nb)); nb += 1
            for r in stage_ranks:
                units = []
                for kind, m in stream:
                    u = []
                    if kind == "F":
                        if stage > 0:
                            u.append(Job(
                                name=f"recv_fwd.mb{m}", kind="p2p",
                                dur=p2p_time, lane="comm", mb=m,
                                gid=f"p2p-f-mb{m}-{stage-1}-{stage}",
                                peers=(per_stage_rank[stage - 1], r)))
                        u.extend(_leaf_jobs_fwd(chunk, stage, m))
                        if stage < pp - 1:
                            u.append(Job(
                                name=f"send_fwd.mb{m}", kind="p2p",
                                dur=p2p_time, lane="comm", mb=m,
                                gid=f"p2p-f-mb{m}-{stage}-{stage+1}",
                                peers=(r, per_stage_rank[stage + 1])))
                    else:
                        if stage < pp - 1:
                            u.append(Job(
                                name=f"recv_bwd.mb{m}", kind="p2p",
                                dur=p2p_time, lane="comm", mb=m,
                                gid=f"p2p-b-mb{m}-{stage+1}-{stage}",
                                peers=(per_stage_rank[stage + 1], r)))
                        u.extend(_leaf_jobs_bwd(chunk, stage, m))
                        if stage > 0:
                            u.append(Job(
                                name=f"send_bwd.mb{m}", kind="p2p",
                                dur=p2p_time, lane="comm", mb=m,
                                gid=f"p2p-b-mb{m}-{stage}-{stage-1}",
                                peers=(r, per_stage_rank[stage - 1])))
                    units.append(u)
                jobs[r].extend(_assemble_units(units, r, s.pp_comm_async))
        self._append_optimizer(jobs, ranks, per_stage_rank)
        return jobs

    def build_interleaved(self) -> Dict[int, List[Job]]:
        """Sync-VPP replay: Megatron interleaved schedule table over the
        per-virtual-chunk models (reference parity:
        pipeline_schedule.py:97-715)."""
        from ..perf.vpp import chunk_id_of, mb_id_of

        s = self.perf.strategy
        pp, mbc = s.pp_size, s.micro_batch_num
        vp = s.interleaving_size
        assert self.perf.vchunks is not None
        p2p_size = get_pp_p2p_comm_size(s, self.perf.model_config)
        p2p_time = self.system.compute_net_op_time(
            "p2p", p2p_size, 2, net=s.pp_net, comm_stage="pp", strategy=s)
        ranks = self.sim_ranks()
        jobs: Dict[int, List[Job]] = {r: [] for r in ranks}
        per_stage_rank = {i: get_pp_stage_representative_rank(i, s)
                          for i in range(pp)}
        nv = pp * vp
        total = mbc * vp
        for stage in range(pp):
            r = per_stage_rank[stage]
            warm = min((pp - stage - 1) * 2 + (vp - 1) * pp, total)
            stream = [("F", k) for k in range(warm)]
            nf, nb = warm, 0
            while nb < total:
                if nf < total:
                    stream.append(("F", nf)); nf += 1
                stream.append(("B", nb)); nb += 1
            units = []
            for kind, k in stream:
                fwd = kind == "F"
                c = chunk_id_of(k, pp, vp, fwd)
                m = mb_id_of(k, pp, vp)
                v = c * pp + stage
                chunk = self.perf.vchunks[stage][c]
                u = []
                if fwd:
                    if v > 0:
                        src = per_stage_rank[stage - 1 if stage > 0 else pp - 1]
                        u.append(Job(
                            name=f"recv_fwd.v{v}.mb{m}", kind="p2p",
                            dur=p2p_time, lane="comm", mb=m,
                            gid=f"p2p-f-mb{m}-v{v-1}-v{v}", peers=(src, r)))
                    u.extend(_leaf_jobs_fwd(chunk, stage, m * vp + c))
                    if v < nv - 1:
                        dst = per_stage_rank[stage + 1 if stage < pp - 1 else 0]
                        u.append(Job(
                            name=f"send_fwd.v{v}.mb{m}", kind="p2p",
                            dur=p2p_time, lane="comm", mb=m,
                            gid=f"p2p-f-mb{m}-v{v}-v{v+1}", peers=(r, dst)))
                else:
                    if v < nv - 1:
                        src = per_stage_rank[stage + 1 if stage < pp - 1 else 0]
                        u.append(Job(
                            name=f"recv_bwd.v{v}.mb{m}", kind="p2p",
                            dur=p2p_time, lane="comm", mb=m,
                            gid=f"p2p-b-mb{m}-v{v+1}-v{v}", peers=(src, r)))
                    u.extend(_leaf_jobs_bwd(chunk, stage, m * vp + c))
                    if v > 0:
                        dst = per_stage_rank[stage - 1 if stage > 0 else pp - 1]
                        u.append(Job(
                            name=f"send_bwd.v{v}.mb{m}", kind="p2p",
                            dur=p2p_time, lane="comm", mb=m,
                            gid=f"p2p-b-mb{m}-v{v}-v{v-1}", peers=(r, dst)))
                units.append(u)
            jobs[r].extend(_assemble_units(units, r, s.pp_comm_async))
        self._append_optimizer(jobs, ranks, per_stage_rank)
        return jobs

    def _append_optimizer(self, jobs, ranks, per_stage_rank):
        s = self.perf.strategy
        # optimizer tail (DP collectives + adam traffic)
        pp = s.pp_size
        for stage in range(pp):
            for r in ([per_stage_rank[stage]] if self.merge_lanes else
                      [x for x in ranks
                       if x // (s.world_size // pp) == stage]):
                dp_t = self.perf._compute_dp_time(stage)
                if s.overlap_grad_reduce and dp_t > 0:
                    # exposed tail only (reduce overlaps last-mb backward)
                    ci = self.perf.chunks[stage].get_cost_info()
                    dp_t = max(0.0, dp_t - (ci.bwd_compute_time
                                            + ci.bwd_net_exposed_time))
                if dp_t > 0:
                    jobs[r].append(Job(name="dp_grad_sync", kind="comm",
                                       dur=dp_t, lane="comm",
                                       gid=f"dp-{stage}", mb=-1))
                jobs[r].append(Job(name="optimizer.adam", kind="optim",
                                   dur=self.perf._compute_optim_time(stage),
                                   mb=-1))
