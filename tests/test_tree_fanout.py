"""Tree-mode SPMD fan-out (reference: tree topology at >=100 workers,
fanout 50): with the threshold lowered, a 4-pod service fans out
coordinator -> 2 children -> grandchild and every rank still reports
(env knobs KT_TREE_THRESHOLD / KT_TREE_FANOUT)."""
import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "assets", "summer"))
os.environ["KT_LOCAL_MODE"] = "true"
os.environ["KT_USERNAME"] = "treetest"
# pods inherit the test env; the coordinator's supervisor module reads
# these at ITS import inside the pod process
os.environ["KT_TREE_THRESHOLD"] = "3"
os.environ["KT_TREE_FANOUT"] = "2"

import kubetorch_amd as kt  # noqa: E402
from tests.assets.summer import summer as summer_mod  # noqa: E402

pytestmark = [pytest.mark.flaky_retry, pytest.mark.minimal]


@pytest.mark.timeout(300)
def test_tree_fanout_reaches_every_rank():
    f = kt.fn(summer_mod.rank_env).to(
        kt.Compute(cpus=1).distribute("pytorch", workers=4, num_proc=1,
                                      quorum_timeout=60))
    try:
        results = f(kt_timeout=180)
        assert isinstance(results, list) and len(results) == 4, results
        assert sorted(r["rank"] for r in results) == [0, 1, 2, 3]
        assert all(r["world_size"] == 4 for r in results)
        # repeat call: tree relays stay correct on warm pods
        results2 = f(kt_timeout=180)
        assert sorted(r["rank"] for r in results2) == [0, 1, 2, 3]
    finally:
        f.teardown()
