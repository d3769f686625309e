"""Context parallelism for the trainer — all three priced modes are
executable (the reference only models them):

* "a2a" (Ulysses): all-to-all scatters heads / gathers sequence around
  flash attention (simulator cp_comm_type="a2a", reference
  dense_module.py:1158-1338). Contiguous shards only.
* "all_gather": K/V gathered to the full sequence, q stays seq-sharded,
  positional causal mask (the mode the reference prices but raises
  NotImplementedError on).
* "ring": K/V blocks circulate over p2p with online-LSE block
  accumulation — absent in the reference entirely.

Shard assignment is "contiguous" or "zigzag" (chunk pairs {c, 2cp-1-c}
for balanced causal load; all_gather/ring only).

Gradient semantics: cp ranks see the same batch but different seq
slices, so parameter grads are averaged over the dp*cp group — the
trainer's reducer spans dp_cp. Process-group layout matches
core/utils.get_rank_group: cp strided by tp (consecutive when tp=1);
composes with DP, TP(+SP), EP, PP and ZeRO-1 (see tests/test_*cp*).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

_CP_GROUPS = {}


def get_cp_groups(cp_size, tp_size=1):
    """cp group = cp_size ranks strided by tp within each tp*cp block
    (Megatron tp-cp-dp rank order, core/utils.get_rank_group). With
    tp_size=1 this is cp_size consecutive ranks.
    Returns (cp_group, cp_rank)."""
    if cp_size <= 1 or not dist.is_initialized():
        return None, 0
    key = (cp_size, tp_size, dist.get_world_size())
    if key not in _CP_GROUPS:
        world = dist.get_world_size()
        block = cp_size * tp_size
        assert world % block == 0
        groups = {}
        for start in range(0, world, block):
            for off in range(tp_size):
                ranks = [start + off + i * tp_size for i in range(cp_size)]
                g = dist.new_group(ranks)
                for r in ranks:
                    groups[r] = g
        _CP_GROUPS[key] = groups
    groups = _CP_GROUPS[key]
    r = dist.get_rank()
    return groups[r], (r // tp_size) % cp_size


def _a2a_exchange(chunks, group):
    """all_to_all of equal-shaped stacked chunks ([cp, ...] tensor).
    gloo has no all_to_all: fall back to all_gather + select."""
    cp = dist.get_world_size(group)
    out = torch.empty_like(chunks)
    backend = dist.get_backend(group)
    if backend == "nccl":
        dist.all_to_all_single(out, chunks.contiguous(), group=group)
    else:
        gathered = [torch.empty_like(chunks) for _ in range(cp)]
        dist.all_gather(gathered, chunks.contiguous(), group=group)
        me = dist.get_rank(group)
        for j in range(cp):
            out[j] = gathered[j][me]
    return out


def _scatter_head_gather_seq(x, group):
    """[B, s, H, d] -> [B, s*cp, H/cp, d] (s local seq, H full heads)."""
    cp = dist.get_world_size(group)
    B, s, H, d = x.shape
    Hc = H // cp
    # [cp, B, s, Hc, d]: chunk i = heads [i*Hc, (i+1)*Hc)
    chunks = x.view(B, s, cp, Hc, d).permute(2, 0, 1, 3, 4).contiguous()
    out = _a2a_exchange(chunks, group)
    # out[j] = seq slice j of my head block -> concat along seq
    return out.permute(1, 0, 2, 3, 4).reshape(B, cp * s, Hc, d)


def _scatter_seq_gather_head(x, group):
    """[B, S, Hc, d] -> [B, S/cp, Hc*cp, d] (inverse of the above)."""
    cp = dist.get_world_size(group)
    B, S, Hc, d = x.shape
    s = S // cp
    # [cp, B, s, Hc, d]: chunk j = seq slice j
    chunks = x.view(B, cp, s, Hc, d).permute(1, 0, 2, 3, 4).contiguous()
    out = _a2a_exchange(chunks, group)
    # out[i] = my seq slice of head block i -> concat along heads
    return out.permute(1, 2, 0, 3, 4).reshape(B, s, cp * Hc, d)


class _HeadScatterSeqGather(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _scatter_head_gather_seq(x, group)

    @staticmethod
    def backward(ctx, dy):
        return _scatter_seq_gather_head(dy.contiguous(), ctx.group), None


class _SeqScatterHeadGather(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _scatter_seq_gather_head(x, group)

    @staticmethod
    def backward(ctx, dy):
        return _scatter_head_gather_seq(dy.contiguous(), ctx.group), None


def cp_pre_attention(x, group):
    """q/k/v [B, s, H, d] -> [B, S, H/cp, d] before flash attention."""
    if group is None:
        return x
    return _HeadScatterSeqGather.apply(x, group)


def cp_post_attention(x, group):
    """o [B, S, H/cp, d] -> [B, s, H, d] after flash attention."""
    if group is None:
        return x
    return _SeqScatterHeadGather.apply(x, group)


# ---- kv all-gather mode (cp_comm_type="all_gather") ----------------------
# The simulator prices this mode (ops/dense.py CoreAttention else-branch:
# ag(k+v) fwd, ag + rs in bwd); the reference models the comm but raises
# NotImplementedError in its flops path (dense_module.py:1521-1524) — here
# it is executable end to end. Each rank keeps its q shard and attends to
# the full gathered K/V with an offset-causal mask.


class _AllGatherSeq(torch.autograd.Function):
    """[B, s, Hkv, d] -> [B, s*cp, Hkv, d] along seq; backward sums the
    full-length gradient over cp and returns this rank's slice (the
    reduce_scatter of the cost model, expressed backend-portably)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        cp = dist.get_world_size(group)
        gathered = [torch.empty_like(x) for _ in range(cp)]
        dist.all_gather(gathered, x.contiguous(), group=group)
        return torch.cat(gathered, dim=1)

    @staticmethod
    def backward(ctx, dy):
        group = ctx.group
        cp = dist.get_world_size(group)
        me = dist.get_rank(group)
        dy = dy.contiguous()
        dist.all_reduce(dy, group=group)
        s = dy.shape[1] // cp
        return dy[:, me * s:(me + 1) * s].contiguous(), None


def cp_allgather_kv(x, group):
    if group is None:
        return x
    return _AllGatherSeq.apply(x, group)


# ---- ring mode (cp_comm_type="ring") -------------------------------------
# Ring attention (blockwise SDP with K/V blocks circulating over p2p and
# online log-sum-exp accumulation) — ABSENT in the reference (SURVEY §2.2
# lists it as not implemented). q stays seq-sharded; each of the cp-1
# ring steps passes the K/V block to the next rank, so peak memory holds
# ONE remote block instead of the all_gather mode's full sequence, and on
# xGMI every hop is an independent point-to-point link.


class _RingPass(torch.autograd.Function):
    """Send x to the next cp rank, receive the previous rank's block.
    Backward reverses the flow (the received block's grad belongs to the
    sender). Deadlock-free via even/odd send/recv ordering (gloo-safe)."""

    @staticmethod
    def _exchange(x, group, to_next):
        cp = dist.get_world_size(group)
        me = dist.get_rank(group)
        ranks = dist.get_process_group_ranks(group)
        nxt = ranks[(me + 1) % cp]
        prv = ranks[(me - 1) % cp]
        dst, src = (nxt, prv) if to_next else (prv, nxt)
        out = torch.empty_like(x)
        x = x.contiguous()
        if me % 2 == 0:
            dist.send(x, dst, group=group)
            dist.recv(out, src, group=group)
        else:
            dist.recv(out, src, group=group)
            dist.send(x, dst, group=group)
        return out

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _RingPass._exchange(x, group, to_next=True)

    @staticmethod
    def backward(ctx, dy):
        return _RingPass._exchange(dy.contiguous(), ctx.group,
                                   to_next=False), None


# ---- shard assignment ----------------------------------------------------
# "contiguous": rank c owns tokens [c*S/cp, (c+1)*S/cp) — simple, but the
# causal mask loads the last rank ~2x the first. "zigzag": 2*cp chunks,
# rank c owns chunks {c, 2cp-1-c} (Megatron CP load balancing) — every
# rank sees the same causal work. a2a (Ulysses) requires contiguous
# (its seq-gather reassembles in rank order for the causal flash kernel).


def cp_shard_slices(S, cp, rank, zigzag):
    if not zigzag:
        s = S // cp
        return [slice(rank * s, (rank + 1) * s)]
    h = S // (2 * cp)
    a, b = rank, 2 * cp - 1 - rank
    return [slice(a * h, (a + 1) * h), slice(b * h, (b + 1) * h)]


def cp_positions(S, cp, rank, zigzag, device):
    """Global positions of this rank's shard, concatenated in shard
    order (int32, length S/cp)."""
    parts = [torch.arange(sl.start, sl.stop, device=device,
                          dtype=torch.int32)
             for sl in cp_shard_slices(S, cp, rank, zigzag)]
    return torch.cat(parts) if len(parts) > 1 else parts[0]


def cp_slice_batch(t, S, cp, rank, zigzag, dim=-1):
    """Slice a [..., S] token/label tensor to this rank's shard."""
    parts = [t.index_select(dim, torch.arange(sl.start, sl.stop,
                                              device=t.device))
             for sl in cp_shard_slices(S, cp, rank, zigzag)]
    return parts[0].contiguous() if len(parts) == 1 else torch.cat(
        parts, dim=dim).contiguous()


def masked_sdp(q, k, v, qpos, kpos):
    """Math SDP with an arbitrary-position causal mask (key kpos[j]
    visible to query qpos[i] iff kpos[j] <= qpos[i]). fp32 softmax; GQA
    via head repetition. Rows with no visible key return 0."""
    B, s, H, d = q.shape
    Hkv = k.shape[2]
    rep = H // Hkv
    kx = k.repeat_interleave(rep, dim=2) if rep > 1 else k
    vx = v.repeat_interleave(rep, dim=2) if rep > 1 else v
    scores = torch.einsum("bqhd,bkhd->bhqk", q.float(), kx.float())
    scores *= d ** -0.5
    scores = scores.masked_fill(kpos[None, None, None, :]
                                > qpos[None, None, :, None],
                                float("-inf"))
    p = torch.softmax(scores, dim=-1)
    out = torch.einsum("bhqk,bkhd->bqhd", p, vx.float())
    return out.to(q.dtype)


def ring_attention_pos(q, k, v, group, cp_rank, qpos, zigzag):
    """Position-aware ring attention (zigzag or contiguous shards):
    blocks circulate with deterministically reconstructed key positions;
    online-LSE accumulation in fp32."""
    cp = dist.get_world_size(group)
    B, s, H, d = q.shape
    Hkv = k.shape[2]
    dv = v.shape[-1]
    rep = H // Hkv
    S_full = s * cp
    qf = q.float()
    num = torch.zeros(B, H, s, dv, device=q.device)
    den = torch.zeros(B, H, s, device=q.device)
    m_run = torch.full((B, H, s), float("-inf"), device=q.device)
    blk = torch.cat([k, v], dim=-1)
    qmax = int(qpos.max())
    for t in range(cp):
        src = (cp_rank - t) % cp
        if t > 0:
            blk = _RingPass.apply(blk, group)
        kpos = cp_positions(S_full, cp, src, zigzag, q.device)
        if int(kpos.min()) > qmax:
            continue       # block entirely in the causal future
        blk_k, blk_v = blk[..., :d], blk[..., d:]
        kx = (blk_k.repeat_interleave(rep, dim=2) if rep > 1 else blk_k)
        vx = (blk_v.repeat_interleave(rep, dim=2) if rep > 1 else blk_v)
        scores = torch.einsum("bqhd,bkhd->bhqk", qf, kx.float()) * d ** -0.5
        scores = scores.masked_fill(kpos[None, None, None, :]
                                    > qpos[None, None, :, None],
                                    float("-inf"))
        m_blk = scores.amax(dim=-1)
        m_new = torch.maximum(m_run, m_blk)
        # fully-masked rows keep m=-inf until a visible block arrives
        alpha = torch.exp((m_run - m_new).nan_to_num(0.0))
        p = torch.exp((scores - m_new[..., None]).nan_to_num(float("-inf")))
        num = num * alpha[..., None] + torch.einsum(
            "bhqk,bkhd->bhqd", p, vx.float())
        den = den * alpha + p.sum(dim=-1)
        m_run = m_new
    out = (num / den[..., None]).permute(0, 2, 1, 3)
    out = out + 0.0 * blk.float().sum()
    return out.to(q.dtype)
