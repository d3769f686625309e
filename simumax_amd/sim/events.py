"""Discrete-event simulator core (L6).

Parity target: simumax/core/base_struct.py:35-231,1225-2763 (SimuSystem
event loop, SimuContext, BarrierBackend/P2PBackend rendezvous, Com op
hierarchy) — re-designed: jobs are flat per-rank lists produced by
sim.schedule.PpSchedule from the analytic chunks' leaf costs; the loop is
a greedy multi-lane scheduler with collective rendezvous and deadlock
diagnostics.

Every executed job emits a LogEvent consumed by sim.trace (Chrome trace)
and sim.memory (allocator timeline).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional


@dataclass
class MemDelta:
    """Memory effect of a job: applied to the rank's allocator timeline."""

    alloc_bytes: float = 0.0        # persists after the job (cache token)
    free_bytes: float = 0.0         # released when the job completes
    transient_bytes: float = 0.0    # peak-only workspace during the job
    token_key: str = ""             # alloc/free pairing key (FIFO checked)


@dataclass
class Job:
    name: str                      # e.g. "stage0.layer3.fc1"
    kind: str                      # fwd | bwd | recompute | optim | comm | p2p
    dur: float                     # ms (compute jobs); comm dur priced already
    lane: str = "comp"             # comp | comm
    # comm rendezvous
    gid: Optional[str] = None      # group id; all participants must arrive
    peers: Optional[tuple] = None  # participating ranks (None = local)
    mb: int = -1                   # microbatch index
    mem: Optional[MemDelta] = None
    call_stack: str = ""
    overlap: bool = False          # comm overlapped with compute: occupies
                                   # only the comm lane (CommEvent.overlap)


@dataclass
class LogEvent:
    rank: int
    name: str
    kind: str
    lane: str
    start: float
    end: float
    mb: int
    call_stack: str = ""
    wait_start: float = 0.0   # time the job became head-of-lane
    mem: object = None        # the job's MemDelta (memory replay)
    gid: Optional[str] = None  # rendezvous id (trace flow arrows)


class DeadlockError(RuntimeError):
    pass


class SimuSystem:
    """Greedy multi-rank, two-lane-per-rank event loop with collective
    rendezvous. Reference parity: SimuSystem.simu base_struct.py:1385-1538
    (incl. the rich deadlock diagnostics)."""

    def __init__(self, jobs_per_rank: Dict[int, List[Job]]):
        self.jobs = jobs_per_rank
        self.log: List[LogEvent] = []

    def run(self) -> float:
        ranks = sorted(self.jobs)
        ptr = {r: 0 for r in ranks}
        # comp = compute stream, comm = serialized sync-collective stream,
        # async = overlapped-collective stream (own HIP stream/communicator,
        # does not contend with sync comm — CommEvent.overlap)
        lane_t = {r: {"comp": 0.0, "comm": 0.0, "async": 0.0} for r in ranks}
        # rendezvous state: gid -> {rank: arrival_time}
        arrivals: Dict[str, Dict[int, float]] = {}
        done: Dict[str, float] = {}  # gid -> completion time

        total = sum(len(j) for j in self.jobs.values())
        executed = 0
        while executed < total:
            progressed = False
            for r in ranks:
                while ptr[r] < len(self.jobs[r]):
                    job = self.jobs[r][ptr[r]]
                    if job.gid is None:
                        start = lane_t[r][job.lane]
                        end = start + job.dur
                        lane_t[r][job.lane] = end
                        if job.lane == "comm":
                            lane = "async" if job.overlap else "comm"
                            # issued when compute reaches this point; each
                            # stream serializes its own transfers
                            start = max(lane_t[r]["comp"], lane_t[r][lane])
                            end = start + job.dur
                            lane_t[r][lane] = end
                            if not job.overlap:
                                # sync collective: blocks compute too
                                lane_t[r]["comp"] = end
                        self.log.append(LogEvent(r, job.name, job.kind, job.lane,
                                                 start, end, job.mb, job.call_stack,
                                                 start, job.mem))
                        ptr[r] += 1
                        executed += 1
                        progressed = True
                        continue
                    # async p2p (strategy.pp_comm_async): the receiver
                    # posted its irecv one unit ahead; the wait job pays
                    # only the remaining transfer time
                    if job.kind == "p2p_post_recv":
                        t = lane_t[r]["comp"]
                        done.setdefault(job.gid + "#rposted", t)
                        self.log.append(LogEvent(r, job.name, "p2p", "comm",
                                                 t, t, job.mb, job.call_stack,
                                                 t, None, gid=job.gid))
                        ptr[r] += 1
                        executed += 1
                        progressed = True
                        continue
                    if job.kind == "p2p_wait":
                        key = job.gid + "#posted"
                        if key not in done:
                            break  # sender has not posted yet
                        rpost = done.get(job.gid + "#rposted",
                                         lane_t[r]["comp"])
                        # the transfer runs on the network from the moment
                        # both sides have posted; the wait only blocks the
                        # receiver until the data is ready
                        xfer_start = max(done[key], rpost)
                        ready = xfer_start + job.dur
                        arrive = lane_t[r]["comp"]
                        start = xfer_start
                        end = max(ready, arrive)
                        lane_t[r]["comm"] = max(lane_t[r]["comm"], end)
                        lane_t[r]["comp"] = max(lane_t[r]["comp"], end)
                        self.log.append(LogEvent(r, job.name, "p2p", "comm",
                                                 start, end, job.mb,
                                                 job.call_stack, arrive,
                                                 job.mem, gid=job.gid))
                        ptr[r] += 1
                        executed += 1
                        progressed = True
                        continue
                    # p2p: eager send (sender posts and proceeds; receiver
                    # pays the transfer after the send is posted) — matches
                    # the analytic recurrence F_dep = f_end[prev] + p2p and
                    # is deadlock-free for 1F1B steady state
                    if job.kind == "p2p" and job.peers and r == job.peers[0] \
                            and job.peers[1] in self.jobs and job.peers[0] != job.peers[1]:
                        post = max(lane_t[r].values())
                        a = arrivals.setdefault(job.gid, {})
                        a[r] = post
                        done.setdefault(job.gid + "#posted", post)
                        self.log.append(LogEvent(r, job.name, "p2p", job.lane,
                                                 post, post, job.mb,
                                                 job.call_stack, post, job.mem,
                                                 gid=job.gid))
                        ptr[r] += 1
                        executed += 1
                        progressed = True
                        continue
                    if job.kind == "p2p" and job.peers and r == job.peers[1] \
                            and job.peers[0] in self.jobs and job.peers[0] != job.peers[1]:
                        key = job.gid + "#posted"
                        if key not in done:
                            break  # wait for the send post
                        arrive = max(lane_t[r].values())
                        start = max(arrive, done[key])
                        end = start + job.dur
                        lane_t[r]["comm"] = max(lane_t[r]["comm"], end)
                        lane_t[r]["comp"] = max(lane_t[r]["comp"], end)
                        self.log.append(LogEvent(r, job.name, "p2p", job.lane,
                                                 start, end, job.mb,
                                                 job.call_stack, arrive, job.mem,
                                                 gid=job.gid))
                        ptr[r] += 1
                        executed += 1
                        progressed = True
                        continue
                    # collective rendezvous
                    peers = job.peers or (r,)
                    sim_peers = tuple(p for p in peers if p in self.jobs)
                    if len(sim_peers) <= 1:
                        # peers not simulated (lane-merged): local cost
                        lane = ("async" if job.overlap and job.lane == "comm"
                                else job.lane)
                        start = max(lane_t[r][lane], lane_t[r]["comp"])
                        end = start + job.dur
                        lane_t[r][lane] = end
                        if not (job.overlap and job.lane == "comm"):
                            lane_t[r]["comp"] = max(lane_t[r]["comp"], end)
                        self.log.append(LogEvent(r, job.name, job.kind, job.lane,
                                                 start, end, job.mb, job.call_stack,
                                                 start, job.mem))
                        ptr[r] += 1
                        executed += 1
                        progressed = True
                        continue
                    if job.gid in done:
                        end = done[job.gid]
                        arrive = arrivals[job.gid][r]
                        self.log.append(LogEvent(r, job.name, job.kind, job.lane,
                                                 end - job.dur, end, job.mb,
                                                 job.call_stack, arrive, job.mem,
                                                 gid=job.gid))
                        lane = ("async" if job.overlap and job.lane == "comm"
                                else job.lane)
                        lane_t[r][lane] = max(lane_t[r][lane], end)
                        if job.lane == "comm" and not job.overlap:
                            # sync comm also blocks the compute lane
                            lane_t[r]["comp"] = max(lane_t[r]["comp"], end)
                        ptr[r] += 1
                        executed += 1
                        progressed = True
                        continue
                    # arrive and block
                    a = arrivals.setdefault(job.gid, {})
                    if r not in a:
                        if job.overlap and job.lane == "comm":
                            # issued at the compute front; queues on the
                            # overlap stream only
                            a[r] = max(lane_t[r]["comp"], lane_t[r]["async"])
                        else:
                            # sync: the rank arrives once its sync lanes
                            # reach this job
                            a[r] = max(lane_t[r]["comp"], lane_t[r]["comm"])
                        progressed = True
                    if all(p in a for p in sim_peers):
                        done[job.gid] = max(a.values()) + job.dur
                        progressed = True
                        continue  # retry same job (now completable)
                    break  # blocked on peers
            if not progressed:
                blocked = {
                    r: (self.jobs[r][ptr[r]].name, self.jobs[r][ptr[r]].gid)
                    for r in ranks if ptr[r] < len(self.jobs[r])
                }
                raise DeadlockError(
                    f"simulation deadlock; blocked heads per rank: {blocked}; "
                    f"pending rendezvous: "
                    f"{ {g: sorted(a) for g, a in arrivals.items() if g not in done} }"
                )
        return max(max(t.values()) for t in lane_t.values()) if ranks else 0.0
