"""Intra-node shuffle exchange: RCCL all-to-all-v over xGMI, replacing the
reference's HTTP fetch (Shuffle/ShuffleScheduler/FetcherOrderedGrouped —
SURVEY §3b/§8e).  Partition p is owned by rank p % world_size; the per-pair
byte counts come from the spill index triples (the ShuffleHeader-equivalent
metadata, ShuffleHeader.java:82-106), exchanged first as a size matrix
(all_gather); the bytes then move as grouped P2P send/recv over the 7
pairwise xGMI links (see _move_a2av — torch's all_to_all_single is avoided:
it silently truncates >1 GB single-peer portions on this RCCL).

Planning functions are torch-free and covered by CPU (gloo) tests; the tensor
movement uses whatever backend the process group has (nccl=RCCL on the GPU
box, gloo in CPU tests)."""
from dataclasses import dataclass
from typing import List


def owner_of(partition: int, world: int) -> int:
    return partition % world


def parts_for_dest(P: int, world: int, dest: int) -> List[int]:
    return [p for p in range(P) if p % world == dest]


@dataclass
class SendPlan:
    """Partition segments ordered by (dest rank, partition asc)."""
    order: List[int]          # partition ids in send order
    byte_splits: List[int]    # per-dest byte counts
    rec_splits: List[int]     # per-dest record counts
    seg_bytes: List[int]      # per ordered segment
    seg_recs: List[int]


def plan_send(rec_ranges, byte_ranges, world: int) -> SendPlan:
    """rec_ranges/byte_ranges: [P+1] prefix arrays from
    tzs_sorter_sorted_columnar (partition p = [r[p], r[p+1]))."""
    P = len(rec_ranges) - 1
    order, seg_bytes, seg_recs = [], [], []
    byte_splits, rec_splits = [], []
    for d in range(world):
        b = r = 0
        for p in parts_for_dest(P, world, d):
            order.append(p)
            sb = int(byte_ranges[p + 1] - byte_ranges[p])
            sr = int(rec_ranges[p + 1] - rec_ranges[p])
            seg_bytes.append(sb)
            seg_recs.append(sr)
            b += sb
            r += sr
        byte_splits.append(b)
        rec_splits.append(r)
    return SendPlan(order, byte_splits, rec_splits, seg_bytes, seg_recs)


_P2P_CHUNK = 256 << 20  # bytes per P2P op: far from the >1 GB RCCL
                        # all_to_all_single truncation (see _move_a2av)


def _move_a2av(send, recv, in_splits, out_splits, group):
    """All-to-all-v as grouped P2P send/recv over xGMI (the north star's
    `ncclGroupStart; ncclSend/ncclRecv` form) with the self-portion moved by
    a direct device copy.  Replaces torch's all_to_all_single, which was
    measured to SILENTLY truncate single-peer portions beyond ~1 GB on this
    RCCL (a 1.76 GB world=1 exchange delivered exactly the first half;
    reproduced with a pure-torch loop — gpurun_out/r2_dbg4).  Ops are
    chunked at 256 MB; both sides chunk identically so P2P matching holds.
    element_size-aware: splits are in ELEMENTS of the tensors' dtype."""
    import torch
    import torch.distributed as dist
    world = dist.get_world_size(group)
    me = dist.get_rank(group)
    esz = send.element_size()
    soff = [0]
    for v in in_splits:
        soff.append(soff[-1] + v)
    roff = [0]
    for v in out_splits:
        roff.append(roff[-1] + v)
    if in_splits[me] != out_splits[me]:
        raise RuntimeError("self split mismatch")
    if in_splits[me]:
        recv[roff[me]:roff[me + 1]].copy_(send[soff[me]:soff[me + 1]])
    if world == 1:
        return
    chunk_elems = max(_P2P_CHUNK // esz, 1)

    def chunks(t, lo, hi):
        while lo < hi:
            c = min(hi - lo, chunk_elems)
            yield t[lo:lo + c]
            lo += c

    backend = dist.get_backend(group)
    if backend == "nccl":
        ops = []
        for peer in range(world):
            if peer == me:
                continue
            for c in chunks(send, soff[peer], soff[peer + 1]):
                ops.append(dist.P2POp(dist.isend, c, peer, group=group))
            for c in chunks(recv, roff[peer], roff[peer + 1]):
                ops.append(dist.P2POp(dist.irecv, c, peer, group=group))
        if ops:
            for req in dist.batch_isend_irecv(ops):
                req.wait()
    else:
        # gloo (CPU tests): true async send/recv; post receives first
        reqs = []
        for peer in range(world):
            if peer == me:
                continue
            for c in chunks(recv, roff[peer], roff[peer + 1]):
                reqs.append(dist.irecv(c, peer, group=group))
        for peer in range(world):
            if peer == me:
                continue
            for c in chunks(send, soff[peer], soff[peer + 1]):
                reqs.append(dist.isend(c, peer, group=group))
        for r in reqs:
            r.wait()


def exchange(plan: SendPlan, send_data, send_reclen, send_klen, group=None,
             nparts=None):
    """all-to-all-v of (data bytes, per-record lengths+klens).  send_* are
    torch tensors already laid out in plan order (uint8/i32/i32).

    Exactly TWO data-path collectives per step: one all_to_all_single for the
    record bytes and one for the per-record (reclen ‖ klen) lengths packed
    into a single i32 tensor (per-dest chunk = reclens then klens).  The
    size matrix rides ONE small all_gather up front (the 24B index triples'
    counts — SURVEY §8e); when ``nparts`` is given the per-partition record
    counts ride the same all_gather and the received records' original
    partition ids are reconstructed here (what exchange_parts previously did
    with its own collective).

    Returns (recv_data, recv_reclen, recv_klen) or, with nparts,
    (recv_data, recv_reclen, recv_klen, recv_parts)."""
    import torch
    import torch.distributed as dist
    world = dist.get_world_size(group)
    me = dist.get_rank(group)
    dev = send_data.device
    in_b = plan.byte_splits
    in_r = plan.rec_splits
    P = int(nparts) if nparts is not None else 0
    counts = [0] * P
    if P:
        for i, p in enumerate(plan.order):
            counts[p] = plan.seg_recs[i]
    my_meta = torch.tensor(in_b + in_r + counts, dtype=torch.int64, device=dev)
    all_meta = [torch.empty_like(my_meta) for _ in range(world)]
    dist.all_gather(all_meta, my_meta, group=group)
    all_meta = [m.tolist() for m in all_meta]
    out_b = [int(all_meta[src][me]) for src in range(world)]
    out_r = [int(all_meta[src][world + me]) for src in range(world)]
    # collective 1: packed lengths — per-dest chunk [reclen_d ‖ klen_d]
    total_in_r = sum(in_r)
    total_out_r = sum(out_r)
    send_len = torch.empty(2 * total_in_r, dtype=torch.int32, device=dev)
    pos = rpos = 0
    for d in range(world):
        r = in_r[d]
        if r:
            send_len[pos:pos + r] = send_reclen[rpos:rpos + r]
            send_len[pos + r:pos + 2 * r] = send_klen[rpos:rpos + r]
        pos += 2 * r
        rpos += r
    recv_len = torch.empty(2 * total_out_r, dtype=torch.int32, device=dev)
    _move_a2av(send_len, recv_len, [2 * x for x in in_r],
               [2 * x for x in out_r], group)
    recv_reclen = torch.empty(total_out_r, dtype=torch.int32, device=dev)
    recv_klen = torch.empty(total_out_r, dtype=torch.int32, device=dev)
    pos = rpos = 0
    for src in range(world):
        r = out_r[src]
        if r:
            recv_reclen[rpos:rpos + r] = recv_len[pos:pos + r]
            recv_klen[rpos:rpos + r] = recv_len[pos + r:pos + 2 * r]
        pos += 2 * r
        rpos += r
    # collective 2: the record bytes
    recv_data = torch.empty(sum(out_b), dtype=torch.uint8, device=dev)
    _move_a2av(send_data, recv_data, in_b, out_b, group)
    # per-source chunk boundaries (lets reduce_merge ingest each source's
    # chunk as one pre-sorted segment — no re-sort on the reduce side)
    plan.recv_rec_splits = out_r
    plan.recv_byte_splits = out_b
    if nparts is None:
        return recv_data, recv_reclen, recv_klen
    # reconstruct received records' partition ids in arrival order
    # (src rank asc, then owned partitions asc — all_to_all concatenation)
    ids, cnts = [], []
    for src in range(world):
        for p in range(P):
            c = int(all_meta[src][2 * world + p])
            if p % world == me and c:
                ids.append(p)
                cnts.append(c)
    if not ids:
        recv_parts = torch.empty(0, dtype=torch.int32, device=dev)
    else:
        recv_parts = torch.repeat_interleave(
            torch.tensor(ids, dtype=torch.int32, device=dev),
            torch.tensor(cnts, dtype=torch.int64, device=dev))
    return recv_data, recv_reclen, recv_klen, recv_parts


def pack_send_tensors(sorter, plan: SendPlan, device):
    """Build the send tensors from a flushed sorter's columnar view
    (device-to-device copies per partition segment)."""
    import torch
    from ._engine import lib, _ck
    d_data, d_off, d_klen, rec_ranges, byte_ranges = sorter.sorted_columnar()
    total_b = sum(plan.byte_splits)
    total_r = sum(plan.rec_splits)
    send_data = torch.empty(max(total_b, 1), dtype=torch.uint8, device=device)
    send_reclen = torch.empty(max(total_r, 1), dtype=torch.int32, device=device)
    send_klen = torch.empty(max(total_r, 1), dtype=torch.int32, device=device)
    # per-record lengths from the off array: copy off into a torch u64 tensor
    n = rec_ranges[-1]
    off_t = torch.empty(n + 1, dtype=torch.int64, device=device)
    if n >= 0 and d_off.value:
        _ck(lib().tzs_memcpy_d2d(off_t.data_ptr(), d_off, 8 * (n + 1)), "d2d")
    reclen_all = (off_t[1:] - off_t[:-1]).to(torch.int32) if n > 0 else off_t.to(torch.int32)[:0]
    klen_all = torch.empty(max(n, 1), dtype=torch.int32, device=device)
    if n > 0:
        _ck(lib().tzs_memcpy_d2d(klen_all.data_ptr(), d_klen, 4 * n), "d2d")
    bpos = rpos = 0
    for i, p in enumerate(plan.order):
        sb, sr = plan.seg_bytes[i], plan.seg_recs[i]
        if sb:
            _ck(lib().tzs_memcpy_d2d(send_data.data_ptr() + bpos,
                                     d_data.value + int(byte_ranges[p]), sb), "d2d")
        if sr:
            r0 = int(rec_ranges[p])
            send_reclen[rpos:rpos + sr] = reclen_all[r0:r0 + sr]
            send_klen[rpos:rpos + sr] = klen_all[r0:r0 + sr]
        bpos += sb
        rpos += sr
    return send_data[:total_b], send_reclen[:total_r], send_klen[:total_r]


def exchange_parts(plan: SendPlan, P: int, device, group=None):
    """All-gather the per-partition record-count matrix (the 24B index
    triples' record counts — the ShuffleHeader-level metadata) and rebuild
    the per-record PARTITION ids of the records this rank receives, in
    arrival order (src rank asc, then owned partitions asc — matching
    all_to_all_single's concatenation and plan_send's packing order).
    Preserves explicit partitioners (range/LUT) across the exchange; for
    HashPartitioner data it also saves the reduce side a re-hash."""
    import torch
    import torch.distributed as dist
    world = dist.get_world_size(group)
    me = dist.get_rank(group)
    counts = [0] * P
    for i, p in enumerate(plan.order):
        counts[p] = plan.seg_recs[i]
    pc = torch.tensor(counts, dtype=torch.int64, device=device)
    all_pc = [torch.empty_like(pc) for _ in range(world)]
    dist.all_gather(all_pc, pc, group=group)
    ids, cnts = [], []
    for src in range(world):
        src_pc = all_pc[src].tolist()
        for p in range(P):
            if p % world == me and src_pc[p]:
                ids.append(p)
                cnts.append(int(src_pc[p]))
    if not ids:
        return torch.empty(0, dtype=torch.int32, device=device)
    return torch.repeat_interleave(
        torch.tensor(ids, dtype=torch.int32, device=device),
        torch.tensor(cnts, dtype=torch.int64, device=device))


def reduce_merge(conf_factory, recv_data, recv_reclen, recv_klen,
                 recv_parts=None, src_rec_splits=None, src_byte_splits=None):
    """Feed received columnar records into a reduce-side sorter (the
    MergeManager/TezMerger replacement) and flush: the final merged IFile for
    this rank's owned partitions.  recv_parts carries the original partition
    ids; without it the sorter recomputes HashPartitioner placement (wrong
    for explicit partitioners).

    With src_rec_splits/src_byte_splits (exchange() records them on the plan
    as recv_rec_splits/recv_byte_splits), each source rank's chunk — already
    sorted by (partition, key) by the map side — is ingested as one
    PRE-SORTED segment and flush runs the k-way merge-path merge over the
    segments, skipping the reduce-side re-sort (MergeManager admission +
    TezMerger merge, MergeManager.java:423-519, TezMerger.java:466-706).
    Without splits it falls back to the unsorted single-batch path."""
    import torch
    n = int(recv_reclen.numel())
    sorter = conf_factory()
    if n:
        off = torch.zeros(n + 1, dtype=torch.int64, device=recv_data.device)
        torch.cumsum(recv_reclen.to(torch.int64), 0, out=off[1:])
        klen_u32 = recv_klen.contiguous()
        parts_t = None
        if recv_parts is not None and int(recv_parts.numel()) == n:
            parts_t = recv_parts.contiguous()
        if recv_data.is_cuda:
            # the collectives/cumsum above run on torch's stream; the engine
            # reads these tensors on the default stream — without this sync a
            # large all_to_all is still in flight when the composite build
            # reads the tail (observed: one zeroed record at ~1 GB received)
            torch.cuda.synchronize()
        if src_rec_splits is not None:
            r0 = 0
            b0 = 0
            keep = []
            for src, r in enumerate(src_rec_splits):
                if r == 0:
                    continue
                b = (int(src_byte_splits[src]) if src_byte_splits is not None
                     else int(off[r0 + r] - off[r0]))
                seg_off = (off[r0:r0 + r + 1] - off[r0]).contiguous()
                keep.append(seg_off)
                sorter.add_sorted_segment(
                    recv_data.data_ptr() + b0,
                    seg_off.data_ptr(),
                    klen_u32.data_ptr() + 4 * r0,
                    parts_t.data_ptr() + 4 * r0 if parts_t is not None else None,
                    r)
                r0 += r
                b0 += b
            sorter._exchange_keepalive = (recv_data, off, klen_u32, parts_t, keep)
        else:
            parts_ptr = parts_t.data_ptr() if parts_t is not None else None
            sorter.write_batch_device(recv_data.data_ptr(), off.data_ptr(),
                                      klen_u32.data_ptr(), parts_ptr, n)
            sorter._exchange_keepalive = (recv_data, off, klen_u32, parts_t)
    sorter.flush()
    return sorter
