"""End-to-end cluster tests over TCP loopback (BASELINE config #1): keystone
+ workers + client, SHM one-sided fast path, striping, replication,
failover, batch APIs."""
import os
import time

import pytest

import blackbird_amd as bb

from conftest import Cluster

MB = 1 << 20


class TestSingleWorker:
    def test_put_get_roundtrip(self, cluster):
        c = cluster.client()
        data = os.urandom(1024)
        c.put("k1", data)
        assert c.get("k1") == data
        c.close()

    def test_various_sizes(self, cluster):
        c = cluster.client()
        for size in [1, 100, 4096, 65536, 1 * MB, 3 * MB + 7]:
            data = os.urandom(size)
            c.put("s%d" % size, data)
            assert c.get("s%d" % size) == data, size
        c.close()

    def test_checksum_verified_on_get(self, cluster):
        c = cluster.client(verify_checksum_on_get=True)
        data = os.urandom(64 * 1024)
        c.put("k", data)
        assert c.get("k") == data
        # corrupt the object in worker memory behind keystone's back
        pool = cluster.workers[0].pool_descriptors()[0]
        be = cluster.workers[0].backend(pool.pool_id)
        info = cluster.keystone.service().get_workers("k")
        off = info.copies[0].shards[0].offset
        be.write(off, b"\xff" * 16)
        with pytest.raises(Exception, match="CHECKSUM_MISMATCH"):
            c.get("k")
        c.close()

    def test_exists_remove(self, cluster):
        c = cluster.client()
        c.put("k", b"x" * 100)
        assert c.exists("k")
        c.remove("k")
        assert not c.exists("k")
        with pytest.raises(Exception, match="OBJECT_NOT_FOUND"):
            c.get("k")
        c.close()

    def test_batch_put_get(self, cluster):
        c = cluster.client()
        items = [("b%03d" % i, os.urandom(8192)) for i in range(64)]
        statuses = c.batch_put(items)
        assert statuses == [0] * 64
        res = c.batch_get([k for k, _ in items])
        for (k, d), (s, got) in zip(items, res):
            assert s == 0 and got == d
        c.close()

    def test_replace_upsert(self, cluster):
        """cfg.replace atomically overwrites an existing key (no separate
        remove RPC); without the flag a duplicate put still fails."""
        c = cluster.client(verify_checksum_on_get=True)
        c.put("slot", b"v1" * 100)
        assert c.get("slot") == b"v1" * 100
        cfg = bb.PlacementConfig()
        cfg.replace = True
        c.put("slot", b"v2" * 4096, cfg)   # different size too
        assert c.get("slot") == b"v2" * 4096
        # batch upsert
        items = [("slot", b"v3" * 64), ("fresh-upsert", b"new")]
        assert c.batch_put(items, cfg) == [0, 0]
        assert c.get("slot") == b"v3" * 64
        assert c.get("fresh-upsert") == b"new"
        # plain put still rejects duplicates
        with pytest.raises(Exception, match="OBJECT_EXISTS"):
            c.put("slot", b"nope")
        c.close()

    def test_same_size_upsert_is_in_place(self, cluster):
        """A same-size replace reuses the existing placement (no allocator
        churn — offsets stay stable, which keeps client placement caches
        hot); the digest still updates to the new content."""
        c = cluster.client(verify_checksum_on_get=True)
        cfg = bb.PlacementConfig()
        cfg.replace = True
        a = os.urandom(64 * 1024)
        b = os.urandom(64 * 1024)
        c.put("slot2", a, cfg)
        ks = cluster.keystone.service()
        sh1 = ks.get_workers("slot2").copies[0].shards[0]
        d1 = ks.get_workers("slot2").checksum
        c.put("slot2", b, cfg)
        info = ks.get_workers("slot2")
        sh2 = info.copies[0].shards[0]
        assert (sh2.pool_id, sh2.offset) == (sh1.pool_id, sh1.offset)
        assert info.checksum != d1  # digest tracked the new bytes
        assert c.get("slot2") == b
        # different size → re-placed, still correct
        big = os.urandom(256 * 1024)
        c.put("slot2", big, cfg)
        assert c.get("slot2") == big
        c.close()

    def test_batch_partial_failure(self, cluster):
        c = cluster.client()
        c.put("dup", b"first")
        items = [("dup", b"second"), ("fresh", b"ok")]
        statuses = c.batch_put(items)
        assert statuses[0] != 0 and statuses[1] == 0
        assert c.get("dup") == b"first"
        assert c.get("fresh") == b"ok"
        c.close()

    def test_list_objects(self, cluster):
        c = cluster.client()
        for i in range(5):
            c.put("app/item%d" % i, b"x" * (1000 + i))
        c.put("other/one", b"y" * 50)
        rows = c.list_objects("app/")
        assert [k for k, *_ in rows] == ["app/item%d" % i for i in range(5)]
        assert rows[0][1] == 1000 and rows[0][2] == 1  # size, ncopies
        assert rows[0][3] == bb.StorageClass.RAM_CPU
        assert len(c.list_objects()) == 6
        assert len(c.list_objects("app/", limit=2)) == 2
        assert c.list_objects("nope/") == []
        c.close()

    def test_cluster_stats(self, cluster):
        c = cluster.client()
        c.put("k", b"z" * 4096)
        st = c.cluster_stats()
        assert st.num_workers == 1 and st.num_pools == 1
        assert st.num_objects == 1 and st.total_used >= 4096
        ws = c.workers_info()
        assert len(ws) == 1 and ws[0].worker_id == "w0"
        c.close()

    def test_ttl_object(self, cluster):
        c = cluster.client()
        cfg = bb.PlacementConfig()
        cfg.ttl_ms = 200
        c.put("ephemeral", b"gone soon", cfg)
        assert c.exists("ephemeral")
        time.sleep(0.6)
        assert not c.exists("ephemeral")
        c.close()


class TestMultiWorker:
    def test_striping(self, cluster3):
        c = cluster3.client()
        cfg = bb.PlacementConfig()
        cfg.max_workers_per_copy = 3
        data = os.urandom(3 * MB)
        c.put("striped", data, cfg)
        info = cluster3.keystone.service().get_workers("striped")
        assert len(info.copies[0].shards) == 3
        assert len({s.worker_id for s in info.copies[0].shards}) == 3
        assert c.get("striped") == data
        c.close()

    def test_replication_and_failover(self, cluster3):
        c = cluster3.client()
        cfg = bb.PlacementConfig()
        cfg.replication = 2
        data = os.urandom(256 * 1024)
        c.put("redundant", data, cfg)
        info = cluster3.keystone.service().get_workers("redundant")
        workers = {cp.shards[0].worker_id for cp in info.copies}
        assert len(workers) == 2
        # kill the worker holding copy 0
        victim_id = info.copies[0].shards[0].worker_id
        victim = next(w for w in cluster3.workers
                      if any(p.worker_id == victim_id for p in w.pool_descriptors()))
        victim.stop()
        time.sleep(1.5)  # heartbeat TTL 1000ms fires, keystone cleans up
        st = c.cluster_stats()
        assert st.num_workers == 2
        # object still readable from the surviving replica
        assert c.get("redundant") == data
        c.close()

    def test_worker_death_without_replica_loses_object(self, cluster3):
        c = cluster3.client()
        data = os.urandom(64 * 1024)
        c.put("fragile", data)
        info = cluster3.keystone.service().get_workers("fragile")
        victim_id = info.copies[0].shards[0].worker_id
        victim = next(w for w in cluster3.workers
                      if any(p.worker_id == victim_id for p in w.pool_descriptors()))
        victim.stop()
        time.sleep(1.5)
        assert not c.exists("fragile")  # dropped, not stale
        c.close()

    def test_remove_worker_api(self, cluster3):
        c = cluster3.client()
        ws = c.workers_info()
        assert len(ws) == 3
        cluster3.keystone.service().remove_worker(ws[0].worker_id)
        time.sleep(0.3)
        assert len(c.workers_info()) == 2
        c.close()


class TestRepair:
    def test_rereplication_after_worker_death(self):
        """An object with replication=2 loses one copy when its worker dies;
        the repair pass restores the second copy on a surviving worker."""
        cl = Cluster(n_workers=4, pool_bytes=32 * MB)
        try:
            c = cl.client()
            cfg = bb.PlacementConfig()
            cfg.replication = 2
            data = os.urandom(512 * 1024)
            c.put("fixme", data, cfg)
            ks = cl.keystone.service()
            info = ks.get_workers("fixme")
            assert len(info.copies) == 2
            victim_id = info.copies[0].shards[0].worker_id
            victim = next(w for w in cl.workers
                          if any(p.worker_id == victim_id
                                 for p in w.pool_descriptors()))
            victim.stop()
            # heartbeat TTL fires → copy dropped → the gc loop's automatic
            # repair pass restores it (gc_interval is 200 ms in the fixture)
            deadline = time.time() + 8
            while time.time() < deadline:
                info = ks.get_workers("fixme")
                workers_now = {cp.shards[0].worker_id for cp in info.copies}
                if len(info.copies) == 2 and victim_id not in workers_now:
                    break
                time.sleep(0.1)
            info = ks.get_workers("fixme")
            assert len(info.copies) == 2
            workers = {cp.shards[0].worker_id for cp in info.copies}
            assert victim_id not in workers
            assert len(workers) == 2
            assert c.get("fixme") == data
            # both copies independently readable: kill the original survivor
            survivor_id = [w for w in workers][0]
            c.close()
        finally:
            cl.stop()

    def test_repair_waits_for_capacity(self):
        cl = Cluster(n_workers=2, pool_bytes=8 * MB)
        try:
            c = cl.client()
            cfg = bb.PlacementConfig()
            cfg.replication = 2
            c.put("solo", os.urandom(1 * MB), cfg)
            ks = cl.keystone.service()
            w1 = cl.workers[1]
            w1.stop()
            time.sleep(1.5)
            assert len(ks.get_workers("solo").copies) == 1
            # only one worker left → repair cannot place a disjoint copy;
            # must not crash and must not double-place on the same worker
            ks.run_repair_once()
            info = ks.get_workers("solo")
            workers = {cp.shards[0].worker_id for cp in info.copies}
            assert len(info.copies) == len(workers)
            c.close()
        finally:
            cl.stop()


class TestTcpDataPlane:
    def test_force_tcp_batch_roundtrip(self, cluster):
        """force_tcp exercises the framed data protocol end to end, with
        batch puts grouped into one DATA_BATCH_WRITE per worker."""
        c = cluster.client(force_tcp=True)
        items = [("tcp%02d" % i, os.urandom(32 * 1024)) for i in range(32)]
        assert c.batch_put(items) == [0] * 32
        res = c.batch_get([k for k, _ in items])
        for (k, d), (s, got) in zip(items, res):
            assert s == 0 and got == d, k
        # single ops through the same path
        c.put("tcp-single", b"x" * 5000)
        assert c.get("tcp-single") == b"x" * 5000
        c.close()


class TestWorkerRestart:
    def test_no_stale_shm_alias_after_restart(self):
        """A restarted worker reusing the same worker/pool ids must get a
        FRESH shm segment name: a client that had the old pool mapped must
        not silently write into the orphaned segment."""
        cl = Cluster(n_workers=1, pool_bytes=32 * MB)
        try:
            c = cl.client(verify_checksum_on_get=True)
            c.put("pre", os.urandom(64 * 1024))
            assert c.get("pre") is not None  # maps the old segment
            old = cl.workers[0]
            old_shm = old.pool_descriptors()[0].access.shm_name
            old.stop()
            cl.workers.clear()
            wc = bb.WorkerConfig()
            wc.worker_id = "w0"  # SAME ids as the fixture's worker
            wc.coord_endpoint = "127.0.0.1:%d" % cl.coord_server.port
            wc.data_listen_address = "127.0.0.1:0"
            wc.heartbeat_interval_ms = 200
            wc.heartbeat_ttl_ms = 1000
            p = bb.PoolConfig()
            p.pool_id = "pool0_0"  # SAME pool id as the old incarnation
            p.storage_class = bb.StorageClass.RAM_CPU
            p.size_bytes = 32 * MB
            wc.pools = [p]
            w2 = bb.WorkerService(wc)
            w2.initialize()
            w2.start()
            cl.workers.append(w2)
            new_shm = w2.pool_descriptors()[0].access.shm_name
            assert new_shm != old_shm  # fresh incarnation, fresh segment
            time.sleep(1.5)  # old heartbeat expires, new pools registered
            data = os.urandom(128 * 1024)
            c.put("post-restart", data, bb.PlacementConfig())
            assert c.get("post-restart") == data  # digest-verified
            c.close()
        finally:
            cl.stop()


class TestScrub:
    def test_scrub_quarantines_corrupt_copy(self):
        """Background digest scrubbing: silent bit-rot in one replica is
        detected, the corrupt copy quarantined, and repair restores a clean
        one — reads never see bad bytes."""
        cl = Cluster(n_workers=3, pool_bytes=32 * MB)
        try:
            c = cl.client(verify_checksum_on_get=True)
            cfg = bb.PlacementConfig()
            cfg.replication = 2
            data = os.urandom(256 * 1024)
            c.put("scrubbed", data, cfg)
            ks = cl.keystone.service()
            info = ks.get_workers("scrubbed")
            assert len(info.copies) == 2
            # flip bits in copy 0 behind everyone's back
            bad = info.copies[0].shards[0]
            victim = next(w for w in cl.workers
                          if any(p.pool_id == bad.pool_id
                                 for p in w.pool_descriptors()))
            victim.backend(bad.pool_id).write(bad.offset, b"\xde\xad" * 32)
            n = ks.run_scrub_once()
            assert n == 1
            info = ks.get_workers("scrubbed")
            assert len(info.copies) == 1  # corrupt copy dropped
            assert c.get("scrubbed") == data
            # repair pass restores replication with a clean copy
            ks.run_repair_once()
            info = ks.get_workers("scrubbed")
            assert len(info.copies) == 2
            assert c.get("scrubbed") == data
            # a second scrub pass is clean (force by resetting the clock)
            assert ks.run_scrub_once(64) == 0
            # maintenance counters observed the events
            ctr = ks.counters()
            assert ctr["scrub_quarantined"] == 1
            assert ctr["repairs"] >= 1
            c.close()
        finally:
            cl.stop()

    def test_scrub_covers_striped_copies(self):
        """Per-shard digests recorded at put time let the scrubber verify
        STRIPED copies shard by shard (the round-1 scrubber silently skipped
        them). Corrupting ONE shard of a 3-way striped object quarantines
        the copy."""
        cl = Cluster(n_workers=3, pool_bytes=32 * MB)
        try:
            c = cl.client()
            cfg = bb.PlacementConfig()
            cfg.max_workers_per_copy = 3
            cfg.replication = 2
            data = os.urandom(3 * MB)
            c.put("striped-scrub", data, cfg)
            ks = cl.keystone.service()
            info = ks.get_workers("striped-scrub")
            assert len(info.copies) == 2
            assert len(info.copies[0].shards) == 3
            # every shard carries its own standalone digest
            for cp in info.copies:
                for sh in cp.shards:
                    assert sh.digest != 0
            # clean pass first
            assert ks.run_scrub_once() == 0
            # corrupt the MIDDLE shard of copy 0 only
            bad = info.copies[0].shards[1]
            victim = next(w for w in cl.workers
                          if any(p.pool_id == bad.pool_id
                                 for p in w.pool_descriptors()))
            victim.backend(bad.pool_id).write(bad.offset + 100, b"\xbe" * 64)
            assert ks.run_scrub_once(64) == 1
            info = ks.get_workers("striped-scrub")
            assert len(info.copies) == 1  # corrupt striped copy quarantined
            assert c.get("striped-scrub") == data  # survivor serves clean bytes
            c.close()
        finally:
            cl.stop()

    def test_scrub_removes_fully_corrupt_object(self):
        cl = Cluster(n_workers=1, pool_bytes=16 * MB)
        try:
            c = cl.client()
            data = os.urandom(64 * 1024)
            c.put("doomed", data)
            ks = cl.keystone.service()
            info = ks.get_workers("doomed")
            sh = info.copies[0].shards[0]
            cl.workers[0].backend(sh.pool_id).write(sh.offset, b"\x00" * 128)
            assert ks.run_scrub_once() == 1
            assert not c.exists("doomed")  # gone, not silently corrupt
            c.close()
        finally:
            cl.stop()


class TestChaos:
    def test_worker_churn_under_load(self):
        """Fault injection: workers join and die randomly while a client
        keeps putting/getting digest-verified objects with replication=2.
        Every successful get must return exactly the stored bytes; the
        cluster must end consistent (repair restores replicas)."""
        import random
        seed = int(os.environ.get("BB_CHAOS_SEED", "1234"))
        rng = random.Random(seed)
        cl = Cluster(n_workers=4, pool_bytes=64 * MB)
        extra_idx = 4
        stored = {}
        try:
            c = cl.client(verify_checksum_on_get=True)
            cfg = bb.PlacementConfig()
            cfg.replication = 2
            errors = []
            for round_ in range(6):
                # mutate cluster membership
                action = rng.choice(["kill", "add", "none"])
                if action == "kill" and len(cl.workers) > 2:
                    victim = cl.workers.pop(rng.randrange(len(cl.workers)))
                    victim.stop()
                elif action == "add":
                    wc = bb.WorkerConfig()
                    wc.worker_id = "wx%d" % extra_idx
                    extra_idx += 1
                    wc.coord_endpoint = "127.0.0.1:%d" % cl.coord_server.port
                    wc.data_listen_address = "127.0.0.1:0"
                    wc.heartbeat_interval_ms = 200
                    wc.heartbeat_ttl_ms = 1000
                    p = bb.PoolConfig()
                    p.pool_id = "xpool%d" % extra_idx
                    p.storage_class = bb.StorageClass.RAM_CPU
                    p.size_bytes = 64 * MB
                    wc.pools = [p]
                    w = bb.WorkerService(wc)
                    w.initialize()
                    w.start()
                    cl.workers.append(w)
                # traffic
                for i in range(10):
                    key = "chaos-%d-%d" % (round_, i)
                    data = os.urandom(rng.choice([4096, 65536, 256 * 1024]))
                    try:
                        c.put(key, data, cfg)
                        stored[key] = data
                    except Exception:
                        pass  # mid-death placement failures are legal
                    if stored and rng.random() < 0.5:
                        k = rng.choice(list(stored))
                        try:
                            got = c.get(k)
                            assert got == stored[k], k
                        except AssertionError:
                            raise
                        except Exception:
                            pass  # transient NOT_FOUND after total copy loss
                time.sleep(0.4)
            # settle: heartbeat TTLs fire, repair runs (gc 200ms)
            time.sleep(2.5)
            # final audit: everything keystone still advertises must verify
            readable = 0
            for k, v in stored.items():
                try:
                    if not c.exists(k):
                        continue
                    assert c.get(k) == v, k
                    readable += 1
                except Exception as e:
                    if "CHECKSUM_MISMATCH" in str(e):
                        raise
            assert readable > 0
            # replication-2 objects that survived should be back to 2 copies
            ks = cl.keystone.service()
            degraded = 0
            for k in stored:
                if not ks.object_exists(k):
                    continue
                if len(ks.get_workers(k).copies) < 2:
                    degraded += 1
            assert degraded <= len(stored) // 4, degraded
            c.close()
        finally:
            cl.stop()


class TestDiscovery:
    def test_client_bootstraps_from_coordination(self, cluster):
        """A client with only the coordd address finds the registered
        keystone through the service registry (the role etcd played for
        the reference's clients)."""
        o = bb.ClientOptions()
        o.keystone_endpoint = ""
        o.coord_endpoint = "127.0.0.1:%d" % cluster.coord_server.port
        c = bb.Client(o)
        c.connect()
        c.put("via-discovery", b"found you")
        assert c.get("via-discovery") == b"found you"
        c.close()


class TestHostSessions:
    def test_host_session_fast_path(self):
        """The DRAM-tier twin of the GPU batch sessions: after the first
        full batch_put (replace mode, stable buffers), steps ride the token
        fast path — two tiny RPCs around direct memcpys. Data and digests
        stay correct as the buffer CONTENTS change between steps, and
        server-side interference invalidates the session transparently."""
        import numpy as np
        cl = Cluster(n_workers=1, pool_bytes=64 << 20)
        try:
            c = cl.client()
            N, S = 16, 8192
            arrs = [np.zeros(S, np.uint8) for _ in range(N)]
            items = [("hs%02d" % i, arrs[i]) for i in range(N)]
            keys = [k for k, _ in items]
            cfg = bb.PlacementConfig()
            cfg.replace = True
            cfg.checksum = True
            sess = bb.HostPutSession()
            ks = cl.keystone.service()
            for step in range(4):
                blobs = [os.urandom(S) for _ in range(N)]
                for a, b in zip(arrs, blobs):
                    a[:] = np.frombuffer(b, np.uint8)
                st = c.batch_put_session(items, cfg, sess)
                assert st == [0] * N, (step, st[:4])
                res = c.batch_get(keys)
                for i, (s_, got) in enumerate(res):
                    assert s_ == 0 and got == blobs[i], (step, i)
                info = ks.get_workers(keys[5])
                assert info.checksum == bb.core.gpu.checksum_cpu(blobs[5])
            assert sess.active
            assert c.host_session_steps >= 3, c.host_session_steps
            # interference: any placement change bumps the epoch → the next
            # step falls back to the full path, then re-establishes
            ks.put_start("intruder", 4096, bb.PlacementConfig())
            ks.put_complete("intruder", checksum=1)
            ks.remove_object("intruder")
            fast_before = c.host_session_steps
            blobs = [os.urandom(S) for _ in range(N)]
            for a, b in zip(arrs, blobs):
                a[:] = np.frombuffer(b, np.uint8)
            assert c.batch_put_session(items, cfg, sess) == [0] * N
            assert c.host_session_steps == fast_before  # full path this step
            assert c.batch_put_session(items, cfg, sess) == [0] * N
            assert c.host_session_steps == fast_before + 1  # re-established
            res = c.batch_get(keys)
            for i, (s_, got) in enumerate(res):
                assert s_ == 0 and got == blobs[i], i
            c.close()
        finally:
            cl.stop()

    def test_host_session_not_kept_without_replace(self):
        import numpy as np
        cl = Cluster(n_workers=1, pool_bytes=64 << 20)
        try:
            c = cl.client()
            arr = np.zeros(4096, np.uint8)
            sess = bb.HostPutSession()
            cfg = bb.PlacementConfig()  # replace=False
            assert c.batch_put_session([("nk", arr)], cfg, sess) == [0]
            assert not sess.active
            assert c.host_session_steps == 0
            c.close()
        finally:
            cl.stop()

    def test_host_session_replicated(self):
        """replication=2 host batches also ride the session fast path: one
        memcpy per replica per step, shared digest on both shards, and the
        data survives losing either worker."""
        import numpy as np
        cl = Cluster(n_workers=2, pool_bytes=64 << 20)
        try:
            c = cl.client()
            N, S = 8, 16384
            arrs = [np.zeros(S, np.uint8) for _ in range(N)]
            items = [("hr%02d" % i, arrs[i]) for i in range(N)]
            keys = [k for k, _ in items]
            cfg = bb.PlacementConfig()
            cfg.replace = True
            cfg.checksum = True
            cfg.replication = 2
            sess = bb.HostPutSession()
            ks = cl.keystone.service()
            for step in range(3):
                blobs = [os.urandom(S) for _ in range(N)]
                for a, b in zip(arrs, blobs):
                    a[:] = np.frombuffer(b, np.uint8)
                assert c.batch_put_session(items, cfg, sess) == [0] * N
            assert c.host_session_steps >= 2, c.host_session_steps
            info = ks.get_workers(keys[2])
            assert len(info.copies) == 2
            assert {cp.shards[0].worker_id for cp in info.copies} == \
                {"w0", "w1"}
            for cp in info.copies:
                assert cp.shards[0].digest == info.checksum
            assert info.checksum == bb.core.gpu.checksum_cpu(blobs[2])
            # kill either worker: the surviving replica serves the bytes
            cl.workers[0].stop()
            deadline = time.time() + 8
            while time.time() < deadline:
                try:
                    cps = ks.get_workers(keys[2]).copies
                    if all(cp.shards[0].worker_id != "w0" for cp in cps):
                        break
                except Exception:
                    pass
                time.sleep(0.1)
            res = c.batch_get(keys)
            for i, (s_, got) in enumerate(res):
                assert s_ == 0 and got == blobs[i], i
            c.close()
        finally:
            cl.stop()

    def test_host_session_survives_ttl_expiry(self):
        """Objects under a session carry a TTL: when they expire and GC
        reclaims them mid-session, the next step must transparently fall
        back, re-place fresh objects and re-establish — never error."""
        import numpy as np
        cl = Cluster(n_workers=1, pool_bytes=64 << 20)
        try:
            c = cl.client()
            N, S = 8, 8192
            arrs = [np.zeros(S, np.uint8) for _ in range(N)]
            items = [("tt%02d" % i, arrs[i]) for i in range(N)]
            cfg = bb.PlacementConfig()
            cfg.replace = True
            cfg.checksum = True
            cfg.ttl_ms = 300
            sess = bb.HostPutSession()
            ks = cl.keystone.service()
            blobs = [os.urandom(S) for _ in range(N)]
            for a, b in zip(arrs, blobs):
                a[:] = np.frombuffer(b, np.uint8)
            for step in range(3):  # establish + ride
                assert c.batch_put_session(items, cfg, sess) == [0] * N
            assert c.host_session_steps >= 1
            time.sleep(0.5)  # TTL fires (each step restarted it; now lapse)
            ks.run_gc_once()  # reclaims the expired objects
            fast = c.host_session_steps
            assert c.batch_put_session(items, cfg, sess) == [0] * N  # re-place
            assert c.batch_put_session(items, cfg, sess) == [0] * N  # fast again
            assert c.host_session_steps == fast + 1
            res = c.batch_get([k for k, _ in items])
            for i, (s_, got) in enumerate(res):
                assert s_ == 0 and got == blobs[i], i
            c.close()
        finally:
            cl.stop()

    def test_host_session_survives_eviction(self):
        """Watermark eviction may reclaim session objects (they are
        COMMITTED between steps): the next step re-places and re-establishes
        with zero caller-visible errors."""
        import numpy as np
        cs = bb.CoordServer(); cs.start("127.0.0.1", 0)
        ep = "127.0.0.1:%d" % cs.port
        kc = bb.KeystoneConfig()
        kc.listen_address = "127.0.0.1:0"
        kc.coord_endpoint = ep
        kc.gc_interval_ms = 100000
        kc.eviction_high_watermark = 0.5
        kc.eviction_ratio = 0.9
        srv = bb.create_and_start_keystone(kc)
        wc = bb.WorkerConfig(); wc.worker_id = "evw0"; wc.coord_endpoint = ep
        wc.data_listen_address = "127.0.0.1:0"
        p = bb.PoolConfig(); p.pool_id = "evp"
        p.storage_class = bb.StorageClass.RAM_CPU; p.size_bytes = 8 << 20
        wc.pools = [p]
        w = bb.WorkerService(wc); w.initialize(); w.start()
        deadline = time.time() + 5
        while time.time() < deadline and not srv.service().get_memory_pools():
            time.sleep(0.02)
        try:
            o = bb.ClientOptions(); o.keystone_endpoint = srv.endpoint
            c = bb.Client(o); c.connect()
            N, S = 8, 64 * 1024
            arrs = [np.zeros(S, np.uint8) for _ in range(N)]
            items = [("ev%02d" % i, arrs[i]) for i in range(N)]
            cfg = bb.PlacementConfig(); cfg.replace = True; cfg.checksum = True
            sess = bb.HostPutSession()
            blobs = [os.urandom(S) for _ in range(N)]
            for a, b in zip(arrs, blobs):
                a[:] = np.frombuffer(b, np.uint8)
            for _ in range(3):
                assert c.batch_put_session(items, cfg, sess) == [0] * N
            # fill past the watermark with OTHER objects, then evict hard
            filler = [("fill%d" % i, b"\xcc" * (512 * 1024)) for i in range(7)]
            c.batch_put(filler)
            time.sleep(0.01)
            srv.service().run_eviction_once()
            # session objects may be gone now: the next steps must recover
            fast = c.host_session_steps
            assert c.batch_put_session(items, cfg, sess) == [0] * N
            assert c.batch_put_session(items, cfg, sess) == [0] * N
            assert c.host_session_steps >= fast + 1
            res = c.batch_get([k for k, _ in items])
            for i, (s_, got) in enumerate(res):
                assert s_ == 0 and got == blobs[i], i
            c.close()
        finally:
            w.stop(); srv.stop(); srv.service().stop(); cs.stop()

    def test_scrub_clean_between_session_steps(self):
        """The background scrubber re-checksums COMMITTED objects; between
        session steps the token-committed digests must verify clean (the
        commit path records digests the scrubber agrees with)."""
        import numpy as np
        cl = Cluster(n_workers=1, pool_bytes=64 << 20)
        try:
            c = cl.client()
            N, S = 8, 32768
            arrs = [np.zeros(S, np.uint8) for _ in range(N)]
            items = [("sc%02d" % i, arrs[i]) for i in range(N)]
            cfg = bb.PlacementConfig(); cfg.replace = True; cfg.checksum = True
            sess = bb.HostPutSession()
            ks = cl.keystone.service()
            for step in range(3):
                blobs = [os.urandom(S) for _ in range(N)]
                for a, b in zip(arrs, blobs):
                    a[:] = np.frombuffer(b, np.uint8)
                assert c.batch_put_session(items, cfg, sess) == [0] * N
                assert ks.run_scrub_once() == 0, step  # nothing quarantined
            assert c.host_session_steps >= 2  # scrub did not break sessions
            res = c.batch_get([k for k, _ in items])
            for i, (s_, got) in enumerate(res):
                assert s_ == 0 and got == blobs[i], i
            c.close()
        finally:
            cl.stop()

    def test_host_session_survives_compaction(self):
        """Pool compaction migrates committed objects to new offsets (epoch
        bump): an established session must fall back transparently on its
        next step and re-establish on the one after."""
        import numpy as np
        cl = Cluster(n_workers=1, pool_bytes=16 << 20)
        try:
            c = cl.client()
            ks = cl.keystone.service()
            N, S = 4, 256 * 1024
            arrs = [np.zeros(S, np.uint8) for _ in range(N)]
            items = [("cp%02d" % i, arrs[i]) for i in range(N)]
            cfg = bb.PlacementConfig(); cfg.replace = True; cfg.checksum = True
            sess = bb.HostPutSession()
            # fragment the pool: interleave fillers, remove half
            fillers = [("frag%d" % i, b"\x11" * (256 * 1024))
                       for i in range(16)]
            assert all(s == 0 for s in c.batch_put(fillers))
            c.batch_remove([k for k, _ in fillers][::2])
            blobs = [os.urandom(S) for _ in range(N)]
            for a, b in zip(arrs, blobs):
                a[:] = np.frombuffer(b, np.uint8)
            for _ in range(3):
                assert c.batch_put_session(items, cfg, sess) == [0] * N
            assert c.host_session_steps >= 1
            pool_id = ks.get_workers("cp00").copies[0].shards[0].pool_id
            moved = ks.compact_pool(pool_id)  # defragment: objects relocate
            fast = c.host_session_steps
            assert c.batch_put_session(items, cfg, sess) == [0] * N
            assert c.batch_put_session(items, cfg, sess) == [0] * N
            if moved:  # placements changed: exactly one fallback step
                assert c.host_session_steps == fast + 1
            res = c.batch_get([k for k, _ in items])
            for i, (s_, got) in enumerate(res):
                assert s_ == 0 and got == blobs[i], i
            c.close()
        finally:
            cl.stop()

    def test_host_session_sustained_with_live_maintenance(self):
        """500 session steps with the maintenance loops (GC/tiering/repair/
        scrub cadence 200 ms) running CONCURRENTLY: every step fast-paths or
        falls back cleanly, periodic full read-backs stay bit-exact."""
        import numpy as np
        cs = bb.CoordServer(); cs.start("127.0.0.1", 0)
        ep = "127.0.0.1:%d" % cs.port
        kc = bb.KeystoneConfig()
        kc.listen_address = "127.0.0.1:0"
        kc.coord_endpoint = ep
        kc.gc_interval_ms = 200  # maintenance churns during the soak
        srv = bb.create_and_start_keystone(kc)
        wc = bb.WorkerConfig(); wc.worker_id = "skw0"; wc.coord_endpoint = ep
        wc.data_listen_address = "127.0.0.1:0"
        p = bb.PoolConfig(); p.pool_id = "skp"
        p.storage_class = bb.StorageClass.RAM_CPU; p.size_bytes = 64 << 20
        wc.pools = [p]
        w = bb.WorkerService(wc); w.initialize(); w.start()
        deadline = time.time() + 5
        while time.time() < deadline and not srv.service().get_memory_pools():
            time.sleep(0.02)
        try:
            o = bb.ClientOptions(); o.keystone_endpoint = srv.endpoint
            c = bb.Client(o); c.connect()
            B, S = 64, 4096
            arrs = [np.zeros(S, np.uint8) for _ in range(B)]
            items = [("sk%02d" % i, arrs[i]) for i in range(B)]
            keys = [k for k, _ in items]
            cfg = bb.PlacementConfig(); cfg.replace = True; cfg.checksum = True
            sess = bb.HostPutSession()
            for step in range(500):
                if step % 50 == 0:
                    for a in arrs:
                        a[:] = np.frombuffer(os.urandom(S), np.uint8)
                assert c.batch_put_session(items, cfg, sess) == [0] * B, step
                if step % 100 == 0:
                    res = c.batch_get(keys)
                    for i, (s_, got) in enumerate(res):
                        assert s_ == 0 and got == arrs[i].tobytes(), (step, i)
            assert c.host_session_steps >= 450  # overwhelmingly fast-path
            c.close()
        finally:
            w.stop(); srv.stop(); srv.service().stop(); cs.stop()
