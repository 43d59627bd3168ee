"""Unit tests for the streamed merge-tree plan builder
(lzy_amd/channels/treeplan.py): component discovery, schedule
construction, placement, chunking and the fold-refusal paths — all pure
CPU, no dist."""
import math

import pytest

from lzy_amd.channels.transport import EntryMeta
from lzy_amd.channels.treeplan import build_plan, find_components


class FakeCall:
    _n = 0

    def __init__(self, inputs, outputs=1, pair_reduce=None, kwargs=None):
        FakeCall._n += 1
        self.pair_reduce = pair_reduce
        self._inputs = tuple(inputs)
        self.entry_ids = tuple(f"out{FakeCall._n}_{i}" for i in range(outputs))
        self.kwarg_entry_ids = dict(kwargs or {})

    def input_entry_ids(self):
        return self._inputs + tuple(self.kwarg_entry_ids.values())


def tensor_meta(eid, owner, shape=(1024,), dtype="float32", device="cpu"):
    return EntryMeta(
        entry_id=eid, owners={owner}, kind="tensor", shape=shape,
        dtype=dtype, device_type=device,
        nbytes=4 * math.prod(shape),
    )


def make_tree(width, owners=None):
    """width leaves -> pairwise merge tree; returns (calls, metas, leaf ids)."""
    FakeCall._n = 0
    calls = {}
    leaves = [f"leaf{i}" for i in range(width)]
    metas = {
        e: tensor_meta(e, owner=(owners[i] if owners else i))
        for i, e in enumerate(leaves)
    }
    layer = leaves[:]
    while len(layer) > 1:
        nxt = []
        for i in range(0, len(layer) - 1, 2):
            c = FakeCall([layer[i], layer[i + 1]], pair_reduce=(0.5, 0.5))
            calls[f"t{len(calls)}"] = c
            nxt.append(c.entry_ids[0])
        if len(layer) % 2:
            nxt.append(layer[-1])
        layer = nxt
    return calls, metas


def test_find_components_tree_and_singletons():
    calls, _ = make_tree(8)
    comps = find_components(calls)
    assert len(comps) == 1
    assert len(comps[0]) == 7  # 4 + 2 + 1 merges
    # topo: every dep precedes its consumer
    pos = {t: i for i, t in enumerate(comps[0])}
    out_of = {calls[t].entry_ids[0]: t for t in comps[0]}
    for t in comps[0]:
        for e in calls[t].input_entry_ids():
            if e in out_of:
                assert pos[out_of[e]] < pos[t]

    # single pairs are not worth folding
    single, _ = make_tree(2)
    assert find_components(single) == []


def test_find_components_skips_cached_ops():
    """cache=True ops must run op-by-op — the plan bypasses the result
    cache entirely, which would silently break the caching contract."""
    calls, _ = make_tree(4)
    some = next(iter(calls))
    calls[some].cache = True
    comps = find_components(calls)
    # the remaining 2 merges still form a component only if connected;
    # with one level-0 merge removed the root links the other two
    assert all(some not in c for c in comps)


def test_find_components_skips_undeclared_and_malformed():
    FakeCall._n = 0
    calls = {
        "a": FakeCall(["x", "y"]),                             # no pair_reduce
        "b": FakeCall(["x", "y", "z"], pair_reduce=(1, 1)),    # 3 inputs...
    }
    # 3-input op can't be a pair reduce even if declared
    calls["b"]._inputs = ("x", "y", "z")
    assert find_components(calls) == []
    # kwargs disqualify
    FakeCall._n = 0
    calls = {
        "a": FakeCall(["x", "y"], pair_reduce=(1, 1), kwargs={"k": "z"}),
        "c": FakeCall(["w", "v"], pair_reduce=(1, 1)),
    }
    assert find_components(calls) == []


def test_build_plan_8_leaves_pipelines_across_levels():
    calls, metas = make_tree(8)
    [order] = find_components(calls)
    plan = build_plan("p1", order, calls, metas.get,
                      chunk_bytes=1024, cuda_p2p=False)
    assert plan is not None
    assert plan["numel"] == 1024
    # chunking: at least 4 chunks for pipelining
    assert plan["numel"] / plan["chunk_elems"] >= 4
    # every rank owning a leaf participates
    assert plan["participants"] == list(range(8))
    # level-0 pairs land on the even ranks (first-input placement)
    nr = plan["node_rank"]
    level0 = order[:4]
    assert sorted(nr[t] for t in level0) == [0, 2, 4, 6]
    # root lands on rank 0 and receives at every level (3 recv steps)
    root = order[-1]
    assert nr[root] == 0
    r0_nodes = [s for s in plan["steps_by_rank"][0] if s["op"] == "node"]
    assert len(r0_nodes) == 3  # level0, level1, root
    # forwarding: rank 2's level-1 node sends to rank 0 (the root);
    # send_to carries (dst, consumer_topo) for tag derivation
    r2_nodes = [s for s in plan["steps_by_rank"][2] if s["op"] == "node"]
    assert any(any(d == 0 for d, _ in s["send_to"]) for s in r2_nodes)
    # pure senders only leaf_send
    r7 = plan["steps_by_rank"][7]
    assert all(s["op"] == "leaf_send" for s in r7)


def test_build_plan_same_rank_leaves_degenerate_local():
    calls, metas = make_tree(4, owners=[3, 3, 3, 3])
    [order] = find_components(calls)
    plan = build_plan("p2", order, calls, metas.get,
                      chunk_bytes=1 << 20, cuda_p2p=False)
    assert plan is not None
    assert plan["participants"] == [3]
    for s in plan["steps_by_rank"][3]:
        assert s["op"] == "node" and s["remote_src"] is None
        assert not s["send_to"]


def test_build_plan_refuses_mismatched_shapes():
    calls, metas = make_tree(4)
    bad = list(metas)[2]
    metas[bad] = tensor_meta(bad, owner=2, shape=(777,))
    [order] = find_components(calls)
    assert build_plan("p3", order, calls, metas.get,
                      chunk_bytes=1024, cuda_p2p=False) is None


def test_build_plan_refuses_missing_meta_and_bytes_kind():
    calls, metas = make_tree(4)
    [order] = find_components(calls)
    missing = dict(metas)
    gone = list(metas)[0]
    del missing[gone]
    assert build_plan("p4", order, calls, missing.get,
                      chunk_bytes=1024, cuda_p2p=False) is None
    metas[gone] = EntryMeta(entry_id=gone, owners={0}, kind="bytes", nbytes=64)
    assert build_plan("p5", order, calls, metas.get,
                      chunk_bytes=1024, cuda_p2p=False) is None


def test_build_plan_chain_shape():
    """Left-deep chain m1=f(a,b); m2=f(m1,c); m3=f(m2,d): accumulator
    stays on a's rank, each level receives from a distinct leaf owner."""
    FakeCall._n = 0
    calls = {}
    metas = {e: tensor_meta(e, owner=i) for i, e in
             enumerate(["a", "b", "c", "d"])}
    prev = "a"
    for i, leaf in enumerate(["b", "c", "d"]):
        c = FakeCall([prev, leaf], pair_reduce=(1.0, -1.0))
        calls[f"m{i}"] = c
        prev = c.entry_ids[0]
    [order] = find_components(calls)
    plan = build_plan("p6", order, calls, metas.get,
                      chunk_bytes=1024, cuda_p2p=False)
    assert plan is not None
    assert all(plan["node_rank"][t] == 0 for t in order)
    # rank 0 receives from 1, 2, 3 in topo order
    srcs = [s["remote_src"] for s in plan["steps_by_rank"][0]
            if s["op"] == "node"]
    assert srcs == [1, 2, 3]


# ---------------------------------------------------------------------------
# property-based: random DAGs / random trees (hypothesis)
# ---------------------------------------------------------------------------
from hypothesis import given, settings, strategies as st


@settings(max_examples=80, deadline=None)
@given(st.data())
def test_find_components_invariants_random_dags(data):
    """For ARBITRARY mixes of pair-reduce and disqualified ops:
    components are disjoint, every member qualifies (pair_reduce,
    2 inputs, 1 output, no kwargs, not cached), each component has >=2
    members, and member order is topological."""
    FakeCall._n = 0
    n_leaves = data.draw(st.integers(2, 10), label="leaves")
    avail = [f"L{i}" for i in range(n_leaves)]
    calls = {}
    for k in range(data.draw(st.integers(0, 14), label="ops")):
        ins = data.draw(
            st.lists(st.sampled_from(avail), min_size=2, max_size=3),
            label=f"in{k}")
        kind = data.draw(st.sampled_from(
            ["pair", "pair", "pair", "plain", "cached", "kw", "multi"]),
            label=f"kind{k}")
        c = FakeCall(
            ins[:2] if kind != "pair" or len(ins) < 3 else ins[:2],
            outputs=2 if kind == "multi" else 1,
            pair_reduce=(1.0, 1.0) if kind != "plain" else None,
            kwargs={"k": ins[-1]} if kind == "kw" else None,
        )
        if kind == "cached":
            c.cache = True
        calls[f"t{k}"] = c
        avail.extend(c.entry_ids)

    comps = find_components(calls)
    seen = set()
    for comp in comps:
        assert len(comp) >= 2
        assert not (set(comp) & seen)  # disjoint
        seen |= set(comp)
        pos = {t: i for i, t in enumerate(comp)}
        out_of = {}
        for t in comp:
            c = calls[t]
            assert c.pair_reduce is not None
            assert len(c.entry_ids) == 1
            assert not c.kwarg_entry_ids
            assert not getattr(c, "cache", False)
            assert len(c.input_entry_ids()) == 2
            out_of[c.entry_ids[0]] = t
        for t in comp:  # topo order within the component
            for e in calls[t].input_entry_ids():
                if e in out_of:
                    assert pos[out_of[e]] < pos[t]


@settings(max_examples=60, deadline=None)
@given(
    width=st.integers(3, 16),
    world=st.integers(1, 8),
    chunk=st.sampled_from([256, 1024, 4096, 1 << 20]),
    data=st.data(),
)
def test_build_plan_invariants_random_trees(width, world, chunk, data):
    """Plans built from random widths/placements/chunk sizes are
    structurally sound: every member gets a rank inside the
    participant set, every send targets a participant (never itself),
    and chunking covers the tensor in >=1 pieces (>=4 when it fits)."""
    owners = [data.draw(st.integers(0, world - 1), label=f"o{i}")
              for i in range(width)]
    calls, metas = make_tree(width, owners=owners)
    comps = find_components(calls)
    if not comps:  # width<=2 never reaches here (width>=3)
        return
    [order] = comps
    plan = build_plan("pp", order, calls, metas.get,
                      chunk_bytes=chunk, cuda_p2p=False)
    if plan is None:  # builder may refuse (e.g. degenerate placement)
        return
    parts = set(plan["participants"])
    assert set(owners) <= parts
    nr = plan["node_rank"]
    assert set(nr) == set(order)
    assert all(r in parts for r in nr.values())
    n_chunks = -(-plan["numel"] // plan["chunk_elems"])
    assert n_chunks >= 1
    if plan["numel"] >= 4:
        assert n_chunks >= 4 or plan["chunk_elems"] * 4 > plan["numel"]
    for rank, steps in plan["steps_by_rank"].items():
        assert rank in parts
        for s in steps:
            for d, _topo in s.get("send_to", []):
                assert d in parts
                assert d != rank  # no self-sends in the schedule
