"""Property-based binpack invariants (hypothesis).

The ledger is the extender's single source of placement truth between
resyncs — these properties hold for EVERY request sequence:
  1. no GPU is ever overcommitted;
  2. assume/release sequences conserve units exactly;
  3. a granted split always sums to the request and only uses listed GPUs;
  4. multi-GPU splits are minimal (no subset of the chosen set fits);
  5. concurrent assumes never overcommit (thread-safety).
"""

from __future__ import annotations

import threading

from hypothesis import given, settings, strategies as st

from gpushare_amd.extender.binpack import BinpackState, NodeGPUState

NODE = "n"

caps = st.lists(st.integers(min_value=1, max_value=64), min_size=1, max_size=8)
requests = st.lists(st.integers(min_value=1, max_value=96), min_size=1, max_size=40)


def full_mesh(n):
    return [[j for j in range(n) if j != i] for i in range(n)]


@given(caps=caps, reqs=requests)
@settings(max_examples=200, deadline=None)
def test_never_overcommit_and_conserve(caps, reqs):
    bs = BinpackState()
    bs.set_node(NODE, caps, xgmi=full_mesh(len(caps)))
    held = []
    for r in reqs:
        split = bs.assume_multi(NODE, r)
        st_node = bs.nodes[NODE]
        for i, cap in enumerate(caps):
            assert 0 <= st_node.allocated[i] <= cap
        if split is not None:
            assert sum(split.values()) == r
            assert all(0 <= i < len(caps) for i in split)
            assert all(u > 0 for u in split.values())
            held.append(split)
        else:
            # infeasible must mean it truly does not fit
            free_total = sum(st_node.free(i) for i in range(len(caps)))
            assert free_total < r
    for split in held:
        bs.release_multi(NODE, split)
    assert bs.packing()["allocated_units"] == 0


@given(caps=caps, r=st.integers(min_value=1, max_value=96))
@settings(max_examples=200, deadline=None)
def test_split_is_minimal(caps, r):
    node = NodeGPUState(NODE, caps, xgmi=full_mesh(len(caps)))
    split = node.best_fit_multi(r)
    if split is None or len(split) == 1:
        return
    # no proper subset of the chosen GPUs can hold the request
    frees = {i: node.free(i) for i in split}
    total = sum(frees.values())
    biggest = max(frees.values())
    assert total - biggest < r  # dropping any (even the largest) breaks fit


@given(caps=st.lists(st.integers(min_value=8, max_value=64), min_size=2, max_size=8))
@settings(max_examples=50, deadline=None)
def test_concurrent_assume_never_overcommits(caps):
    bs = BinpackState()
    bs.set_node(NODE, caps, xgmi=full_mesh(len(caps)))
    results = []
    lock = threading.Lock()

    def worker():
        while True:
            split = bs.assume_multi(NODE, 5)
            if split is None:
                return
            with lock:
                results.append(split)

    threads = [threading.Thread(target=worker) for _ in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    st_node = bs.nodes[NODE]
    for i, cap in enumerate(caps):
        assert st_node.allocated[i] <= cap
    assert sum(sum(s.values()) for s in results) == st_node.total_allocated


# ----------------------------------------------------------------------- #
# devlist codec: native encoder must be byte-identical to the python oracle
# ----------------------------------------------------------------------- #

ids_strategy = st.lists(
    st.text(
        alphabet="abcdef0123456789-", min_size=1, max_size=40
    ).map(lambda s: s + "-_-0"),
    min_size=0,
    max_size=50,
)


@given(ids=ids_strategy, data=st.data())
@settings(max_examples=100, deadline=None)
def test_native_codec_matches_python_oracle(ids, data):
    from gpushare_amd.device.fakedev import encode_list_python, make_codec

    unhealthy = set(
        data.draw(
            st.lists(
                st.integers(min_value=0, max_value=max(len(ids) - 1, 0)),
                max_size=len(ids),
            )
        )
    ) if ids else set()
    codec = make_codec(ids)
    assert codec.encode(sorted(unhealthy)) == encode_list_python(ids, unhealthy)
