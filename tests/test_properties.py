"""Property-based tests (hypothesis) for the pure-logic cores: KV wire
protocol roundtrips, PS shard layout, rank math, webdataset splits."""

import string

from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from tf_yarn_amd._task_commons import compute_rank
from tf_yarn_amd.parallel.ps import shard_parameters
from tf_yarn_amd.pytorch.web_dataset import split_by_rank

KEYS = st.text(alphabet=string.ascii_letters + string.digits + ":/_-",
               min_size=1, max_size=64)
BLOBS = st.binary(min_size=0, max_size=4096)


# the test deletes every key it writes, so reusing one live server
# across generated inputs is sound
@settings(max_examples=30, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(pairs=st.lists(st.tuples(KEYS, BLOBS), min_size=1, max_size=20,
                      unique_by=lambda kv: kv[0]))
def test_kv_roundtrip_arbitrary_keys_and_blobs(kv_client, pairs):
    for k, v in pairs:
        kv_client.put(k, v)
    for k, v in pairs:
        assert kv_client.get(k) == v
    for k, _ in pairs:
        kv_client.delete(k)
    for k, _ in pairs:
        assert kv_client.get(k) is None


@settings(max_examples=100, deadline=None)
@given(shapes=st.lists(
    st.lists(st.integers(1, 64), min_size=0, max_size=3),
    min_size=1, max_size=30),
    n_shards=st.integers(1, 8))
def test_shard_parameters_is_a_partition(shapes, n_shards):
    shapes_t = [tuple(s) for s in shapes]
    shards = shard_parameters(shapes_t, n_shards)
    assert len(shards) == n_shards
    flat = sorted(i for s in shards for i in s)
    assert flat == list(range(len(shapes_t)))  # every param exactly once
    # deterministic: both sides compute the same layout
    assert shards == shard_parameters(shapes_t, n_shards)


@settings(max_examples=100, deadline=None)
@given(n_tasks=st.integers(1, 8), n_procs=st.integers(1, 8))
def test_compute_rank_is_bijective(n_tasks, n_procs):
    ranks = [compute_rank(t, l, n_procs)
             for t in range(n_tasks) for l in range(n_procs)]
    assert sorted(ranks) == list(range(n_tasks * n_procs))


@settings(max_examples=100, deadline=None)
@given(n_items=st.integers(0, 50), world=st.integers(1, 8))
def test_split_by_rank_partitions(n_items, world):
    items = list(range(n_items))
    parts = [split_by_rank(items, r, world) for r in range(world)]
    flat = sorted(x for p in parts for x in p)
    assert flat == items
    # balanced within 1
    lens = [len(p) for p in parts]
    assert max(lens) - min(lens) <= 1


def test_feature_shard_ownership_partitions_features():
    # mirrors the ownership rule in sharded_embedding.py:252
    for world in (1, 2, 3, 4, 8):
        feats = [[f for f in range(26) if f % world == s]
                 for s in range(world)]
        flat = sorted(f for p in feats for f in p)
        assert flat == list(range(26))
        lens = [len(p) for p in feats]
        assert max(lens) - min(lens) <= 1
