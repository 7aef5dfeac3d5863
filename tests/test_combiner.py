"""Combiner (SUM_INT) — SURVEY §8f row 1: fold-by-key during spill and, when
numSpills >= tez.runtime.combine.min.spills, during the final merge
(runCombineProcessor call sites PipelinedSorter.java:602-609,816-821).

CPU tests pin the oracle's fold against in-test Python sums; GPU tests check
engine-vs-oracle byte parity."""
import collections
import random

import numpy as np
import pytest

import oracle as o


def _wordish_pairs(n, nkeys, seed):
    rng = random.Random(seed)
    keys = [b"key%04d" % i for i in range(nkeys)]
    return [(o.serialize_text(keys[rng.randrange(nkeys)]),
             rng.randrange(-100, 1000)) for _ in range(n)]


def _ser(pairs):
    return [(k, int(v).to_bytes(4, "big", signed=True)) for k, v in pairs]


def test_oracle_combiner_spill_sums():
    P = 4
    pairs = _wordish_pairs(2000, 37, seed=3)
    data, off, klen = o.build_records(_ser(pairs))
    res = o.spill(data, off, klen, P, key_type=o.KEY_TEXT, comparator=o.CMP_TEXT,
                  combiner=1)
    idx = o.index_decode(res["index"], P)
    got = {}
    for p in range(P):
        st, raw, cl = idx[p]
        if not cl:
            continue
        for k, v, same in o.ifile_read(res["data"][st:st + cl], with_header=True):
            assert not same  # folded keys are unique
            got[k] = int.from_bytes(v, "big", signed=True)
    want = collections.Counter()
    for k, v in pairs:
        want[k] += v
    # java int wrap semantics
    assert got == {k: ((v + 2**31) % 2**32) - 2**31 for k, v in want.items()}


def test_oracle_combiner_merge_gate():
    """3 spills with the combiner: folded at merge; result equals the fold of
    the union."""
    P = 2
    allp = []
    spills = []
    for s_ in range(3):
        pairs = _wordish_pairs(500, 11, seed=10 + s_)
        allp += pairs
        d, f, k = o.build_records(_ser(pairs))
        spills.append(o.spill(d, f, k, P, key_type=o.KEY_TEXT,
                              comparator=o.CMP_TEXT, combiner=1))
    merged = o.final_merge(spills, P, comparator=o.CMP_TEXT, combiner=1)
    d2, f2, k2 = o.build_records(_ser(allp))
    union = o.spill(d2, f2, k2, P, key_type=o.KEY_TEXT, comparator=o.CMP_TEXT,
                    combiner=1)
    assert merged["data"] == union["data"]
    assert merged["index"] == union["index"]


@pytest.mark.gpu
class TestCombinerGpu:
    @pytest.fixture(scope="class")
    def engine(self):
        import __graft_entry__
        __graft_entry__.build()
        import tez_amd
        if not tez_amd.device_available():
            pytest.skip("no GPU")
        return tez_amd

    def test_single_spill_parity(self, engine):
        P = 8
        pairs = _wordish_pairs(5000, 61, seed=21)
        conf = engine.make_conf(P, key_type=engine.KEY_TEXT,
                                comparator=engine.CMP_TEXT, combiner=1)
        s = engine.Sorter(conf)
        for k, v in _ser(pairs):
            s.write(k, v, -1)
        s.flush()
        got, gidx = s.output()
        s.close()
        d, f, kl = o.build_records(_ser(pairs))
        want = o.spill(d, f, kl, P, key_type=o.KEY_TEXT, comparator=o.CMP_TEXT,
                       combiner=1)
        assert gidx == o.index_decode(want["index"], P)
        assert got == want["data"]

    def test_merge_gate_parity(self, engine):
        """3 spills -> combine at merge; 2 spills -> no combine at merge
        (min.spills=3), both matching the oracle."""
        for nspill, gate in ((3, True), (2, False)):
            P = 4
            conf = engine.make_conf(P, key_type=engine.KEY_TEXT,
                                    comparator=engine.CMP_TEXT, combiner=1)
            s = engine.Sorter(conf)
            spills = []
            for k_ in range(nspill):
                pairs = _wordish_pairs(800, 17, seed=30 + k_)
                for kk, vv in _ser(pairs):
                    s.write(kk, vv, -1)
                s.spill()
                d, f, kl = o.build_records(_ser(pairs))
                spills.append(o.spill(d, f, kl, P, key_type=o.KEY_TEXT,
                                      comparator=o.CMP_TEXT, combiner=1))
            s.flush()
            got, gidx = s.output()
            s.close()
            want = o.final_merge(spills, P, comparator=o.CMP_TEXT,
                                 combiner=1 if gate else 0)
            assert gidx == o.index_decode(want["index"], P), f"nspill={nspill}"
            assert got == want["data"], f"nspill={nspill}"

    def test_wordcount_with_combiner(self, engine):
        from tez_amd.ordered_output import OrderedPartitionedKVOutput
        from tez_amd.ordered_input import OrderedGroupedKVInput
        P = 4
        props = {"tez.runtime.key.class": "org.apache.hadoop.io.Text",
                 "tez.runtime.value.class": "org.apache.hadoop.io.IntWritable",
                 "tez.runtime.combiner.class": "sum"}
        words = ("a bb ccc dddd eeeee ffffff g hh iii jjjj".split())
        rng = random.Random(8)
        doc = [words[rng.randrange(len(words))] for _ in range(30000)]
        out = OrderedPartitionedKVOutput(P, props, unique_id="attempt_c0").start()
        w = out.get_writer()
        for word in doc:
            w.write(word.encode(), 1)
        out.close()
        counted = {}
        for p in range(P):
            inp = OrderedGroupedKVInput(p, props)
            seg, _ = out.segment(p)
            inp.add_segment(seg)
            inp.start()
            for key, vals in inp.get_reader():
                # map-side combined: exactly one value per key here
                assert len(vals) == 1
                counted[key.decode()] = int.from_bytes(vals[0], "big")
        assert counted == dict(collections.Counter(doc))
