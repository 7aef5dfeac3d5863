"""End-to-end plumbing on GPU: the OrderedWordCount shape (BASELINE config 1 /
tez-examples OrderedWordCount.java) through the plugin surface —
OrderedPartitionedKVOutput -> events -> local DISK_DIRECT segments ->
OrderedGroupedKVInput -> grouped reader; plus the world_size=1 exchange path
(pack -> all_to_all -> reduce merge) on device tensors."""
import collections
import random

import pytest

import oracle as o

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def engine():
    import __graft_entry__
    __graft_entry__.build()
    import tez_amd
    if not tez_amd.device_available():
        pytest.skip("no GPU")
    return tez_amd


WORDS = ("the quick brown fox jumps over lazy dog alpha beta gamma delta "
         "epsilon zeta eta theta").split()


def test_ordered_wordcount_e2e(engine):
    from tez_amd.ordered_output import OrderedPartitionedKVOutput
    from tez_amd.ordered_input import OrderedGroupedKVInput
    from tez_amd import events as ev

    P = 4
    props = {"tez.runtime.key.class": "org.apache.hadoop.io.Text",
             "tez.runtime.value.class": "org.apache.hadoop.io.IntWritable"}
    rng = random.Random(5)
    docs = [[WORDS[rng.randrange(len(WORDS))] for _ in range(20000)]
            for _ in range(2)]

    outputs = []
    for m, doc in enumerate(docs):
        out = OrderedPartitionedKVOutput(P, props, unique_id=f"attempt_m{m}").start()
        w = out.get_writer()
        for word in doc:
            w.write(word.encode(), 1)
        evs = out.close()
        assert isinstance(evs[1], ev.CompositeDataMovementEvent)
        # empty-partition info in the DME payload must match the index
        d = ev.parse_dme_payload(evs[1].payload)
        empt = {p for p in range(P) if out._index[p][1] <= 6}
        assert d["empty_partitions"] == empt
        assert d["path_component"] == f"attempt_m{m}"
        outputs.append(out)

    counted = {}
    for p in range(P):
        inp = OrderedGroupedKVInput(p, props)
        for out in outputs:
            seg, _raw = out.segment(p)
            inp.add_segment(seg)
        inp.start()
        prev = None
        for key, vals in inp.get_reader():
            word = key.decode()
            # partition placement: HashPartitioner over utf8 content
            assert (o.hash_bytes(key) & 0x7FFFFFFF) % P == p
            if prev is not None:
                assert prev < key  # sorted, grouped => strictly increasing
            prev = key
            counted[word] = counted.get(word, 0) + sum(
                int.from_bytes(v, "big") for v in vals)

    want = collections.Counter(w for doc in docs for w in doc)
    assert counted == dict(want)


def test_exchange_world1_on_device(engine):
    """world_size=1 degenerate exchange exercises pack_send_tensors (d2d
    segment copies, reclen/klen tensors) and reduce_merge on real device
    tensors; output must equal the direct single-sorter path."""
    import torch
    import torch.distributed as dist
    import os
    from tez_amd import exchange as ex

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29521")
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1)
    torch.cuda.set_device(0)
    device = torch.device("cuda", 0)

    P, n = 8, 30000
    conf = engine.make_conf(P)
    d, off, kl, part = engine.generate(seed=99, n=n, kind=0, klen=16, vlen=24,
                                       conf=conf)
    # direct path
    s_direct = engine.Sorter(engine.make_conf(P))
    s_direct.write_batch_device(d, off, kl, None, n)
    s_direct.flush()
    want_data, want_idx = s_direct.output()
    s_direct.close()
    # exchange path
    m = engine.Sorter(engine.make_conf(P))
    m.write_batch_device(d, off, kl, None, n)
    m.flush()
    _, _, _, rr, br = m.sorted_columnar()
    plan = ex.plan_send(rr, br, 1)
    sd, srl, skl = ex.pack_send_tensors(m, plan, device)
    rd, rrl, rkl, rparts = ex.exchange(plan, sd, srl, skl, nparts=P)
    # product path: each source chunk ingested as a PRE-SORTED segment
    # (tzs_sorter_add_sorted_segment; no reduce-side re-sort)
    red = ex.reduce_merge(lambda: engine.Sorter(engine.make_conf(P)), rd, rrl, rkl,
                          rparts, src_rec_splits=plan.recv_rec_splits,
                          src_byte_splits=plan.recv_byte_splits)
    got_data, got_idx = red.output()
    red.close()
    # legacy unsorted-batch path must agree too
    red2 = ex.reduce_merge(lambda: engine.Sorter(engine.make_conf(P)), rd, rrl, rkl,
                           ex.exchange_parts(plan, P, device))
    got2_data, got2_idx = red2.output()
    red2.close()
    m.close()
    assert got2_idx == got_idx
    assert got2_data == got_data
    engine.free_device(d, off, kl, part)
    assert got_idx == want_idx
    assert got_data == want_data
    dist.destroy_process_group()


def test_pipelined_shuffle_events_and_segments(engine):
    """final merge disabled: per-spill DMEs with spill_id/last_event
    (PipelinedSorter.java:374-385,709-726); reduce side merges the per-spill
    partition segments."""
    from tez_amd.ordered_output import OrderedPartitionedKVOutput
    from tez_amd.ordered_input import OrderedGroupedKVInput
    from tez_amd import events as ev

    P = 2
    props = {"tez.runtime.key.class": "org.apache.hadoop.io.Text",
             "tez.runtime.value.class": "org.apache.hadoop.io.IntWritable",
             "tez.runtime.enable.final-merge.in.output": "false"}
    out = OrderedPartitionedKVOutput(P, props, unique_id="attempt_p0").start()
    w = out.get_writer()
    rng = random.Random(7)
    words1 = [WORDS[rng.randrange(len(WORDS))] for _ in range(3000)]
    for word in words1:
        w.write(word.encode(), 1)
    mid_ev = out.spill()
    d0 = ev.parse_dme_payload(mid_ev.payload)
    assert d0["spill_id"] == 0 and d0["last_event"] is False
    assert d0["path_component"] == "attempt_p0_0"
    words2 = [WORDS[rng.randrange(len(WORDS))] for _ in range(3000)]
    for word in words2:
        w.write(word.encode(), 1)
    evs = out.close()
    # VM event + one DME per spill; the last carries last_event=true
    dmes = [e for e in evs if isinstance(e, ev.CompositeDataMovementEvent)]
    assert len(dmes) == 2
    dlast = ev.parse_dme_payload(dmes[-1].payload)
    assert dlast["spill_id"] == 1 and dlast["last_event"] is True

    counted = {}
    for p in range(P):
        inp = OrderedGroupedKVInput(p, props)
        for sid in range(2):
            seg, _ = out.spill_segment(sid, p)
            inp.add_segment(seg)
        inp.start()
        for key, vals in inp.get_reader():
            counted[key.decode()] = counted.get(key.decode(), 0) + sum(
                int.from_bytes(v, "big") for v in vals)
    want = collections.Counter(words1 + words2)
    assert counted == dict(want)
