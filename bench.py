#!/usr/bin/env python3
"""bench.py — BASELINE.json metric: shuffled+sorted KV bytes/sec.

Workload (config.workload): BASELINE.json configs[1] (C2) — the largest
single-GPU config the metric is quoted on: 1e8 records, 16B random unique key
+ 64B value (BytesWritable serialization), 64 partitions, HashPartitioner +
TezBytesComparator.  One step = the full reference-equivalent pass over one
batch: absorb (device-resident input) -> stable radix sort -> IFile emit +
CRC -> spill index; N>1 adds the RCCL all-to-all-v exchange + reduce-side
merge (partitions sharded p % N, weak scaling).  Inputs are generated on
device BEFORE the timed region; no disk I/O in the timed region (spills are
HBM-resident — DESIGN.md §4).

value = whole-job Σ serialized key+value bytes (the reference's
TaskCounter.OUTPUT_BYTES semantics, PipelinedSorter.java:466) ÷ max-over-ranks
wall time per step.  The CPU oracle (kind "port") is timed on a bounded
sample as cpu_baseline; roofline reports the dominant kernel (radix scatter)
against the 8 TB/s gfx950 HBM peak.
"""
import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

# C2 workload constants
KLEN, VLEN, PARTS = 16, 64, 64
REC_SER = 4 + KLEN + 4 + VLEN  # serialized record bytes (BytesWritable k+v)
SEED = 0x7E2C2


def run_step_c3(tez_amd, gen_batches, free_inputs=False):
    """C3: forced multi-spill + k-way merge (BASELINE configs[2]).  Each
    batch is absorbed and spilled (sorted + IFile-emitted), then flush runs
    the 32-way merge into the final output.  At the full 1e9-record size
    (~92 GB payload) the sorter ADOPTS each batch (zero-copy absorb: one
    materialization of the records total, like the reference's collect
    serializing into the sort buffer) and spill streams are transient so
    everything fits the 288 GB HBM."""
    t0 = time.perf_counter()
    conf = tez_amd.make_conf(256, key_type=tez_amd.KEY_TEXT,
                             comparator=tez_amd.CMP_TEXT,
                             discard_spill_streams=1 if free_inputs else 0)
    s = tez_amd.Sorter(conf)
    for d, off, kl, n in gen_batches:
        if free_inputs:
            s.write_batch_device_adopt(d, off, kl, None, n)  # consumes inputs
        else:
            s.write_batch_device(d, off, kl, None, n)
        s.spill()
    t1 = time.perf_counter()
    s.flush()
    t2 = time.perf_counter()
    ctr = s.counters()
    tms = s.times()
    s.close()
    tms["host_absorb_ns"] = int((t1 - t0) * 1e9)
    tms["host_flush_ns"] = int((t2 - t1) * 1e9)
    tms["host_create_ns"] = 0
    tms["host_close_ns"] = 0
    return ctr, tms


def run_step_c5(tez_amd, batches):
    """C5 slice: range-partitioned (TotalOrderPartitioner-style) explicit
    placements through the C-ABI.  One batch = single spill, flush = rename;
    with --spills > 1 each batch is spilled and flush runs the
    explicit-partition k-way merge (the round-1 rc=-22 refusal)."""
    t0 = time.perf_counter()
    conf = tez_amd.make_conf(128)
    s = tez_amd.Sorter(conf)
    for d, off, kl, part, n in batches:
        s.write_batch_device(d, off, kl, part, n)
        if len(batches) > 1:
            s.spill()
    t1 = time.perf_counter()
    s.flush()
    t2 = time.perf_counter()
    ctr = s.counters()
    tms = s.times()
    s.close()
    tms["host_create_ns"] = 0
    tms["host_absorb_ns"] = int((t1 - t0) * 1e9)
    tms["host_flush_ns"] = int((t2 - t1) * 1e9)
    tms["host_close_ns"] = 0
    return ctr, tms


def run_step_single(tez_amd, conf, d, off, kl, n, adopt=False, part=None):
    t0 = time.perf_counter()
    s = tez_amd.Sorter(conf)
    t1 = time.perf_counter()
    if adopt:
        s.write_batch_device_adopt(d, off, kl, part, n)
    else:
        s.write_batch_device(d, off, kl, part, n)
    t2 = time.perf_counter()
    s.flush()
    t3 = time.perf_counter()
    ctr = s.counters()
    tms = s.times()
    s.close()
    t4 = time.perf_counter()
    tms["host_create_ns"] = int((t1 - t0) * 1e9)
    tms["host_absorb_ns"] = int((t2 - t1) * 1e9)
    tms["host_flush_ns"] = int((t3 - t2) * 1e9)
    tms["host_close_ns"] = int((t4 - t3) * 1e9)
    return ctr, tms


def run_step_multi(tez_amd, rank, world, device, d, off, kl, n, adopt=False,
                   part=None, nparts=PARTS):
    from tez_amd import exchange as ex
    conf = tez_amd.make_conf(nparts, world_size=world, rank=rank)
    m = tez_amd.Sorter(conf)
    if adopt:
        m.write_batch_device_adopt(d, off, kl, part, n)
    else:
        m.write_batch_device(d, off, kl, part, n)
    m.flush()
    ctr = m.counters()
    d_data, d_off2, d_klen2, rec_ranges, byte_ranges = m.sorted_columnar()
    plan = ex.plan_send(rec_ranges, byte_ranges, world)
    sd, sr, sk = ex.pack_send_tensors(m, plan, device)
    rd, rrl, rkl, rparts = ex.exchange(plan, sd, sr, sk, nparts=nparts)
    red = ex.reduce_merge(lambda: tez_amd.Sorter(tez_amd.make_conf(nparts)),
                          rd, rrl, rkl, rparts,
                          src_rec_splits=plan.recv_rec_splits,
                          src_byte_splits=plan.recv_byte_splits)
    tms = red.times()
    red.close()
    m.close()
    return ctr, tms


def cpu_baseline_line(sample_records=1_000_000):
    """Oracle (CPU restatement) timed on the same workload shape, bounded
    sample (~10-30 s)."""
    import numpy as np
    import oracle as o
    rng = np.random.default_rng(SEED)
    n = sample_records
    data = np.zeros(n * REC_SER, dtype=np.uint8)
    view = data.reshape(n, REC_SER)
    view[:, 0:4] = np.frombuffer(np.int32(KLEN).byteswap().tobytes(), dtype=np.uint8)
    view[:, 4:4 + KLEN] = rng.integers(0, 256, size=(n, KLEN), dtype=np.uint8)
    # force uniqueness like the device generator: mix the record id
    ids = np.arange(n, dtype=np.uint32).view(np.uint8).reshape(n, 4)
    view[:, 8:12] ^= ids
    view[:, 4 + KLEN:8 + KLEN] = np.frombuffer(
        np.int32(VLEN).byteswap().tobytes(), dtype=np.uint8)
    view[:, 8 + KLEN:] = rng.integers(0, 256, size=(n, VLEN), dtype=np.uint8)
    offs = np.arange(0, REC_SER * (n + 1), REC_SER, dtype=np.uint64)
    klens = np.full(n, 4 + KLEN, dtype=np.uint32)
    t0 = time.perf_counter()
    o.spill(data, offs, klens, PARTS, key_type=o.KEY_BYTES, comparator=o.CMP_TEZBYTES)
    dt1 = time.perf_counter() - t0
    cores = min(os.cpu_count() or 1, PARTS)  # partition-parallel: at most P workers
    t0 = time.perf_counter()
    o.spill_mt(data, offs, klens, PARTS, cores)
    dtm = time.perf_counter() - t0
    return {
        "value": n * REC_SER / dtm,
        "unit": "bytes/s",
        "cores": cores,
        "kind": "port",
        "sample": f"oracle partition-parallel spill on {n} records of the C2 "
                  f"shape ({dtm:.2f}s on {cores} threads; single-threaded "
                  f"tzo_spill: {n * REC_SER / dt1 / 1e6:.0f} MB/s in {dt1:.2f}s)",
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--records", type=int, default=100_000_000,
                    help="total records across all ranks (C2 default 1e8)")
    ap.add_argument("--workload", choices=["c2", "c3", "c4", "c5"], default="c2",
                    help="c3: 1 GPU, Text/Zipf keys, 256 partitions, forced "
                         "spills merged at flush (BASELINE configs[2]; "
                         "--records total, --spills segments). "
                         "c4: 199 partitions with Zipf(1.0)-skewed partition "
                         "SIZES (configs[3]); 2 GiB per rank default; at "
                         "world>1 the skew drives the all-to-all-v. "
                         "c5: TeraSort-shaped 10B key + 90B value, range "
                         "partitions (single-GPU slice of configs[4])")
    ap.add_argument("--spills", type=int, default=None,
                    help="forced spill count (default: 32 for c3, 1 for c5)")
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    ap.add_argument("--no-adopt", action="store_true",
                    help="force the copying absorb path (default: the sorter "
                         "adopts per-step pre-generated inputs zero-copy when "
                         "(warmup+steps) x payload fits comfortably in HBM)")
    ap.add_argument("--force-exchange", action="store_true",
                    help="run the all-to-all-v exchange + reduce-merge path even "
                         "at world_size=1 (bench-code validation)")
    ap.add_argument("--traffic-bytes", type=float, default=2.42e9,
                    help="PMC-measured HBM bytes per dominant-kernel "
                         "(k_onesweep_pass) launch at the default 1e8-record "
                         "workload: 2xFETCH_SIZE + WRITE_SIZE per rocprofv3 "
                         "--pmc with the gfx950 FETCH calibration (profiles/"
                         "r02_pmc_final.txt; 1.01x of the 2.4 GB algorithmic "
                         "bytes). Pass 0 to report null.")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = max(args.gpus, world)
    if args.gpus > 1 and world == 1:
        sys.exit("--gpus N>1 must be launched via torch.distributed.run "
                 "(one rank per GPU); a single-process run would report an "
                 "N-inflated value")

    import __graft_entry__
    __graft_entry__.build()
    import tez_amd

    dist = None
    device = None
    use_exchange = world > 1 or args.force_exchange
    if use_exchange:
        import torch
        import torch.distributed as tdist
        dist = tdist
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29531")
        if not tdist.is_initialized():
            tdist.init_process_group("nccl", rank=rank, world_size=world)

    # WEAK scaling (as reported in the JSON): every rank processes the full
    # per-GPU batch; whole-job bytes grow with N.  (records is per rank.)
    n_local = args.records
    conf = tez_amd.make_conf(PARTS)
    c5_part = None
    adopt = False
    if args.workload == "c5":
        assert n_gpus == 1, "c5 bench line is the single-GPU slice"
        c5conf = tez_amd.make_conf(128)
        if args.records == 100_000_000:
            args.records = 500_000_000  # default C5 slice: 5e8 x 100B = 50 GB
        nsp5 = args.spills or 1
        c5_batches = []
        per5 = args.records // nsp5
        for k in range(nsp5):
            d, off, kl, c5_part = tez_amd.generate(seed=SEED + 5 + 131 * k,
                                                   n=per5, kind=2, klen=10,
                                                   vlen=90, conf=c5conf)
            c5_batches.append((d, off, kl, c5_part, per5))
    elif args.workload == "c3":
        assert n_gpus == 1, "c3 is the single-GPU merge config"
        args.spills = args.spills or 32
        c3conf = tez_amd.make_conf(256, key_type=tez_amd.KEY_TEXT,
                                   comparator=tez_amd.CMP_TEXT)
        per = args.records // args.spills
        gen_batches = []
        for k in range(args.spills):
            d, off, kl, part = tez_amd.generate(seed=SEED + 7 * k, n=per, kind=1,
                                                klen=0, vlen=64, conf=c3conf)
            tez_amd.free_device(part)
            gen_batches.append((d, off, kl, per))
    else:
        # C2/C4: one input set per step so the sorter can ADOPT it zero-copy
        # (one materialization of the records, like the reference's collect
        # serializing into its sort buffer).  Falls back to one shared input
        # + copying absorb when the pre-generated sets would not fit HBM.
        # C4 (configs[3]): 199 partitions with Zipf(1.0)-skewed SIZES via the
        # generator's inverse-CDF LUT; explicit per-record partitions ride
        # through the C-ABI; 2 GiB per rank by default.
        c4 = args.workload == "c4"
        if c4:
            if args.records == 100_000_000:
                args.records = 20_000_000  # ~2 GiB of 100B records per rank
                n_local = args.records
            klen_w, vlen_w, kind_w, parts_w = 10, 90, 3, 199
            conf = tez_amd.make_conf(parts_w)
        else:
            klen_w, vlen_w, kind_w, parts_w = KLEN, VLEN, 0, PARTS
        rec_ser_w = 4 + klen_w + 4 + vlen_w
        total_sets = args.warmup + args.steps
        # 230 GB cap: pre-generated inputs + ~15 GB step working set must
        # stay under the touched-page capacity of the 288 GB HBM (hipMalloc
        # overcommits silently; first touches past capacity fault at
        # page-migration speed — DESIGN §7a).  21 sets of the C2 shape
        # (driver --steps 20 --warmup 1) = 185 GB: adopt survives it.
        adopt = (not args.no_adopt
                 and total_sets * n_local * rec_ser_w <= 230e9)
        if adopt:
            input_sets = []
            for sidx in range(total_sets):
                d, off, kl, part = tez_amd.generate(
                    seed=SEED + rank + 7919 * sidx, n=n_local, kind=kind_w,
                    klen=klen_w, vlen=vlen_w, conf=conf)
                if not c4:
                    tez_amd.free_device(part)
                    part = None
                input_sets.append((d, off, kl, part))
            set_cursor = [0]
        else:
            d, off, kl, part = tez_amd.generate(seed=SEED + rank, n=n_local,
                                                kind=kind_w, klen=klen_w,
                                                vlen=vlen_w, conf=conf)
            if not c4:
                tez_amd.free_device(part)
                part = None

    def barrier_sync():
        if dist:
            import torch
            torch.cuda.synchronize()
            dist.barrier()
            torch.cuda.synchronize()

    c3_free_inputs = args.workload == "c3" and args.records >= 500_000_000

    def regen_c3():
        out = []
        per = args.records // args.spills
        c3c = tez_amd.make_conf(256, key_type=tez_amd.KEY_TEXT,
                                comparator=tez_amd.CMP_TEXT)
        for k in range(args.spills):
            d2, off2, kl2, part2 = tez_amd.generate(seed=SEED + 7 * k, n=per,
                                                    kind=1, klen=0, vlen=64,
                                                    conf=c3c)
            tez_amd.free_device(part2)
            out.append((d2, off2, kl2, per))
        return out

    def one_step():
        if args.workload == "c5":
            return run_step_c5(tez_amd, c5_batches)
        if args.workload == "c3":
            return run_step_c3(tez_amd, gen_batches, c3_free_inputs)
        if adopt:
            sd, so, sk, spart = input_sets[set_cursor[0]]
            set_cursor[0] += 1
            if use_exchange:
                return run_step_multi(tez_amd, rank, world, device, sd, so, sk,
                                      n_local, adopt=True, part=spart,
                                      nparts=parts_w)
            return run_step_single(tez_amd, conf, sd, so, sk, n_local, adopt=True,
                                   part=spart)
        if use_exchange:
            return run_step_multi(tez_amd, rank, world, device, d, off, kl, n_local,
                                  part=part, nparts=parts_w)
        return run_step_single(tez_amd, conf, d, off, kl, n_local, part=part)

    last_ctr = last_tms = None
    if c3_free_inputs and args.steps > 1:
        print("# c3 at >=5e8 records supports a single timed step; forcing --steps 1",
              flush=True)
        args.steps = 1
    for _ in range(args.warmup):
        last_ctr, last_tms = one_step()
        if c3_free_inputs:
            gen_batches = regen_c3()   # untimed regeneration (pool-backed)
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        last_ctr, last_tms = one_step()
    barrier_sync()
    elapsed = time.perf_counter() - t0
    if dist:
        import torch
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if args.workload == "c3":
        if not c3_free_inputs:
            for d, off, kl, _n in gen_batches:
                tez_amd.free_device(d, off, kl)
    elif args.workload == "c5":
        for d, off, kl, c5_part, _n in c5_batches:
            tez_amd.free_device(d, off, kl, c5_part)
    elif not adopt:
        tez_amd.free_device(d, off, kl)
        if part is not None:
            tez_amd.free_device(part)
    # adopt mode: every input set was consumed (ownership moved to the sorter)

    total_bytes_per_step = last_ctr["output_bytes"] * n_gpus  # whole-job Σ
    ms_per_step = elapsed / args.steps * 1e3
    value = total_bytes_per_step * args.steps / elapsed

    if rank == 0:
        # roofline: dominant kernel = radix scatter; algorithmic bytes per
        # element per launch = 24 (read u64 key + u32 idx, write both)
        dk_ns = max(last_tms["dominant_kernel_ns"], 1)
        dk_elems = last_tms["dominant_kernel_elems"]
        achieved_gbps = 24.0 * dk_elems / dk_ns  # bytes/ns == GB/s
        roofline = {
            "bound": "hbm",
            "achieved": round(achieved_gbps, 1),
            "peak": 8000.0,
            "unit": "GB/s",
            "frac": round(achieved_gbps / 8000.0, 4),
            "traffic": (args.traffic_bytes
                        if args.traffic_bytes and args.records == 100_000_000
                        and world == 1 else None),
        }
        cpu = None
        if not args.skip_cpu_baseline and world == 1 and args.workload == "c2":
            cpu = cpu_baseline_line()
        out = {
            "metric": "shuffled+sorted KV bytes/sec",
            "value": round(value, 1),
            "unit": "bytes/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "u8",
            "data": "synthetic",
            "config": {
                "workload": ("C5 slice: TeraSort 10B key + 90B value, 128 range "
                             "partitions (TotalOrderPartitioner-style)"
                             if args.workload == "c5" else
                             "C3: Text/Zipf keys 4-32B + 64B vals, 256 partitions, "
                             f"{args.spills} spills k-way merged"
                             if args.workload == "c3" else
                             "C4: 100B records, 199 partitions, Zipf(1.0)-skewed "
                             "partition sizes (all-to-all-v shape)"
                             if args.workload == "c4" else
                             "C2: 1e8 rec x (16B unique key + 64B val), 64 partitions,"
                             " BytesWritable/TezBytesComparator, ordered shuffle"),
                "records": args.records,
                "key_bytes": 10 if args.workload in ("c4", "c5") else KLEN,
                "value_bytes": 90 if args.workload in ("c4", "c5") else VLEN,
                "partitions": (199 if args.workload == "c4" else
                               128 if args.workload == "c5" else
                               256 if args.workload == "c3" else PARTS),
                "parallelism": ("shuffle-shard p%" + str(n_gpus))
                               if n_gpus > 1 else "single",
            },
            "roofline": roofline,
            "cpu_baseline": cpu,
            "phase_ms": {k: round(v / 1e6, 2) for k, v in last_tms.items()
                         if k.endswith("_ns")},
            "pool": {k: (round(v / 1e9, 2) if k != "drops" else v)
                     for k, v in tez_amd.pool_stats().items()},
        }
        print(json.dumps(out))
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
