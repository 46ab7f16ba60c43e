"""Collective batch shuffle (feature-store all-to-all, BASELINE config #5 at
N ranks): the full algorithm — want-list exchange, owner-side resolution and
gather, consistent-abort handshake, data all-to-all — runs on the host
loopback exchanger here; RCCL over xGMI is the production transport
(tests/test_gpu.py::TestRcclShuffle, bench.py --shuffle)."""
import os

import pytest

import blackbird_amd as bb


def make_cluster_objects(n, per_rank=8, size=4096):
    return [{f"r{r}o{i}": os.urandom(size) for i in range(per_rank)}
            for r in range(n)]


class TestLoopbackShuffle:
    def test_symmetric_ring(self):
        n = 4
        objs = make_cluster_objects(n)
        want = [[[] for _ in range(n)] for _ in range(n)]
        for r in range(n):
            p = (r + 1) % n
            want[r][p] = [f"r{p}o{i}" for i in range(8)]
        out = bb.core.loopback_shuffle_for_test(objs, want)
        for r in range(n):
            p = (r + 1) % n
            assert out[r][p] == b"".join(objs[p][f"r{p}o{i}"]
                                         for i in range(8))

    def test_full_alltoall_with_self(self):
        n = 3
        objs = make_cluster_objects(n, per_rank=6, size=1000)
        want = [[[f"r{p}o{i}" for i in (0, 3, 5)] for p in range(n)]
                for _ in range(n)]
        out = bb.core.loopback_shuffle_for_test(objs, want)
        for r in range(n):
            for p in range(n):
                assert out[r][p] == b"".join(objs[p][f"r{p}o{i}"]
                                             for i in (0, 3, 5)), (r, p)

    def test_empty_slots(self):
        n = 3
        objs = make_cluster_objects(n, per_rank=2)
        want = [[[] for _ in range(n)] for _ in range(n)]
        want[0][1] = ["r1o0"]  # a single one-directional transfer
        out = bb.core.loopback_shuffle_for_test(objs, want)
        assert out[0][1] == objs[1]["r1o0"]
        assert out[1][0] == b"" and out[2][1] == b""

    def test_uneven_sizes(self):
        n = 2
        objs = [{"a": os.urandom(17), "b": os.urandom(65536)},
                {"c": os.urandom(1), "d": os.urandom(12345)}]
        want = [[[], ["c", "d"]], [["b", "a"], []]]
        out = bb.core.loopback_shuffle_for_test(objs, want)
        assert out[0][1] == objs[1]["c"] + objs[1]["d"]
        assert out[1][0] == objs[0]["b"] + objs[0]["a"]

    def test_missing_key_aborts_all_ranks_consistently(self):
        # one rank cannot serve a requested object: the totals handshake
        # broadcasts the failure, every rank errors out, nobody hangs
        n = 3
        objs = make_cluster_objects(n, per_rank=2)
        want = [[[] for _ in range(n)] for _ in range(n)]
        want[0][1] = ["r1o0", "NO_SUCH_KEY"]
        want[2][1] = ["r1o1"]  # an innocent pair also aborts
        with pytest.raises(Exception):
            bb.core.loopback_shuffle_for_test(objs, want)
