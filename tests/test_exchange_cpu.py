"""CPU (gloo, world_size=2) tests of the exchange: partition->owner plan and
the all-to-all-v movement.  Runs with CPU tensors here; the same
code path runs over nccl(=RCCL) on the GPU box."""
import os

import numpy as np
import pytest

from tez_amd import exchange as ex


def test_plan_send_shapes():
    # P=7, world=3: partitions by owner p%3
    P, world = 7, 3
    rec = [0, 2, 2, 5, 9, 9, 10, 12]      # prefix ranges
    byt = [0, 20, 20, 50, 90, 90, 100, 120]
    plan = ex.plan_send(rec, byt, world)
    assert plan.order == [0, 3, 6, 1, 4, 2, 5]
    assert sum(plan.byte_splits) == 120
    assert sum(plan.rec_splits) == 12
    # dest 0 owns partitions 0,3,6: bytes 20 + 40 + 20 = 80
    assert plan.byte_splits[0] == (20 - 0) + (90 - 50) + (120 - 100)
    assert plan.rec_splits[0] == 2 + 4 + 2


def _worker(rank, world, tmpdir):
    import torch
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29517"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    P = 5
    # fake sorted-columnar data per rank: records "r{rank}p{part}i{i}" grouped
    # by partition
    recs = {p: [f"r{rank}p{p}i{i}".encode() for i in range(rank + 1 + p)]
            for p in range(P)}
    blob = b""
    rec_ranges = [0]
    byte_ranges = [0]
    reclens, klens = [], []
    for p in range(P):
        for r in recs[p]:
            blob += r
            reclens.append(len(r))
            klens.append(1)
        rec_ranges.append(len(reclens))
        byte_ranges.append(len(blob))
    plan = ex.plan_send(rec_ranges, byte_ranges, world)
    # lay out send tensors in plan order
    sd, srl, skl = b"", [], []
    for i, p in enumerate(plan.order):
        b0, b1 = byte_ranges[p], byte_ranges[p + 1]
        r0, r1 = rec_ranges[p], rec_ranges[p + 1]
        sd += blob[b0:b1]
        srl += reclens[r0:r1]
        skl += klens[r0:r1]
    send_data = torch.from_numpy(np.frombuffer(sd, dtype=np.uint8).copy())
    send_reclen = torch.tensor(srl, dtype=torch.int32)
    send_klen = torch.tensor(skl, dtype=torch.int32)
    rd, rrl, rkl = ex.exchange(plan, send_data, send_reclen, send_klen)
    # expected: from each src rank (in rank order), that src's records for MY
    # owned partitions in partition order
    want = b""
    want_lens = []
    for src in range(world):
        srecs = {p: [f"r{src}p{p}i{i}".encode() for i in range(src + 1 + p)]
                 for p in range(P)}
        for p in ex.parts_for_dest(P, world, rank):
            for r in srecs[p]:
                want += r
                want_lens.append(len(r))
    got = bytes(rd.numpy().tobytes())
    assert got == want, f"rank {rank} data mismatch"
    assert rrl.tolist() == want_lens
    assert rkl.tolist() == [1] * len(want_lens)
    # per-record partition reconstruction (explicit-partitioner provenance)
    rparts = ex.exchange_parts(plan, P, rd.device)
    want_parts = []
    for src in range(world):
        for p in ex.parts_for_dest(P, world, rank):
            want_parts += [p] * (src + 1 + p)
    assert rparts.tolist() == want_parts, f"rank {rank} parts mismatch"
    # fused form: identical results with ONE meta all_gather
    # + two all_to_all collectives (packed lengths, data)
    rd2, rrl2, rkl2, rp2 = ex.exchange(plan, send_data, send_reclen, send_klen,
                                       nparts=P)
    assert bytes(rd2.numpy().tobytes()) == want
    assert rrl2.tolist() == want_lens
    assert rkl2.tolist() == [1] * len(want_lens)
    assert rp2.tolist() == want_parts, f"rank {rank} fused parts mismatch"
    # per-source chunk splits (reduce_merge feeds one pre-sorted segment
    # per source rank): counts must tile the received tensors exactly
    want_splits = []
    for src in range(world):
        srecs = {p: src + 1 + p for p in range(P)}
        want_splits.append(sum(srecs[p] for p in ex.parts_for_dest(P, world, rank)))
    assert plan.recv_rec_splits == want_splits
    assert sum(plan.recv_byte_splits) == rd2.numel()
    dist.destroy_process_group()


def test_exchange_gloo_world2():
    import torch.multiprocessing as mp
    mp.spawn(_worker, args=(2, None), nprocs=2, join=True)


def test_exchange_gloo_world4():
    import torch.multiprocessing as mp
    import os
    os.environ["MASTER_PORT"] = "29518"
    mp.spawn(_worker4, args=(4,), nprocs=4, join=True)


def _worker4(rank, world):
    import torch
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ.setdefault("MASTER_PORT", "29518")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    P = 11
    recs = {p: [f"w4r{rank}p{p}i{i}".encode() for i in range((rank + p) % 3 + 1)]
            for p in range(P)}
    blob = b""
    rec_ranges, byte_ranges, reclens, klens = [0], [0], [], []
    for p in range(P):
        for r in recs[p]:
            blob += r
            reclens.append(len(r))
            klens.append(2)
        rec_ranges.append(len(reclens))
        byte_ranges.append(len(blob))
    plan = ex.plan_send(rec_ranges, byte_ranges, world)
    sd, srl, skl = b"", [], []
    for i, p in enumerate(plan.order):
        sd += blob[byte_ranges[p]:byte_ranges[p + 1]]
        srl += reclens[rec_ranges[p]:rec_ranges[p + 1]]
        skl += klens[rec_ranges[p]:rec_ranges[p + 1]]
    rd, rrl, rkl = ex.exchange(
        plan,
        torch.from_numpy(np.frombuffer(sd, dtype=np.uint8).copy()),
        torch.tensor(srl, dtype=torch.int32),
        torch.tensor(skl, dtype=torch.int32))
    want = b""
    for src in range(world):
        srecs = {p: [f"w4r{src}p{p}i{i}".encode() for i in range((src + p) % 3 + 1)]
                 for p in range(P)}
        for p in ex.parts_for_dest(P, world, rank):
            for r in srecs[p]:
                want += r
    assert bytes(rd.numpy().tobytes()) == want, f"rank {rank}"
    dist.destroy_process_group()


def _worker_chunked(rank, world):
    """Tiny _P2P_CHUNK forces every peer portion through many chunked
    isend/irecv ops — the matching order must hold across ranks."""
    import torch
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ.setdefault("MASTER_PORT", "29519")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    ex._P2P_CHUNK_SAVE = ex._P2P_CHUNK
    try:
        ex.__dict__["_P2P_CHUNK"] = 5
        P = 6
        recs = {p: [f"ckr{rank}p{p}i{i}".encode() for i in range(3 + p)]
                for p in range(P)}
        blob = b""
        rec_ranges, byte_ranges, reclens, klens = [0], [0], [], []
        for p in range(P):
            for r in recs[p]:
                blob += r
                reclens.append(len(r))
                klens.append(1)
            rec_ranges.append(len(reclens))
            byte_ranges.append(len(blob))
        plan = ex.plan_send(rec_ranges, byte_ranges, world)
        sd, srl, skl = b"", [], []
        for i, p in enumerate(plan.order):
            sd += blob[byte_ranges[p]:byte_ranges[p + 1]]
            srl += reclens[rec_ranges[p]:rec_ranges[p + 1]]
            skl += klens[rec_ranges[p]:rec_ranges[p + 1]]
        rd, rrl, rkl = ex.exchange(
            plan, torch.from_numpy(np.frombuffer(sd, dtype=np.uint8).copy()),
            torch.tensor(srl, dtype=torch.int32),
            torch.tensor(skl, dtype=torch.int32))
        want = b""
        for src in range(world):
            srecs = {p: [f"ckr{src}p{p}i{i}".encode() for i in range(3 + p)]
                     for p in range(P)}
            for p in ex.parts_for_dest(P, world, rank):
                for r in srecs[p]:
                    want += r
        assert bytes(rd.numpy().tobytes()) == want, f"rank {rank} chunked"
    finally:
        ex.__dict__["_P2P_CHUNK"] = ex._P2P_CHUNK_SAVE
    dist.destroy_process_group()


def test_exchange_gloo_chunked_p2p():
    import torch.multiprocessing as mp
    os.environ["MASTER_PORT"] = "29519"
    mp.spawn(_worker_chunked, args=(2,), nprocs=2, join=True)
