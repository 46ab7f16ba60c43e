"""Coordination service tests (the etcd-equivalent): KV, TTL leases, prefix
scans, watches, CAS and leader election — embedded and over TCP."""
import time

import pytest

import blackbird_amd as bb


class TestInProc:
    def test_put_get_del(self, coord):
        coord.put("/a", "1")
        assert coord.get("/a") == "1"
        coord.delete_("/a")
        with pytest.raises(Exception, match="KEY_NOT_FOUND"):
            coord.get("/a")

    def test_prefix(self, coord):
        for i in range(5):
            coord.put("/p/%d" % i, str(i))
        coord.put("/q/x", "other")
        kvs = coord.get_prefix("/p/")
        assert len(kvs) == 5
        assert sorted(k for k, v in kvs) == ["/p/%d" % i for i in range(5)]

    def test_ttl_expiry(self, coord):
        coord.put("/t", "v", ttl_ms=150)
        assert coord.get("/t") == "v"
        time.sleep(0.5)
        with pytest.raises(Exception, match="KEY_NOT_FOUND"):
            coord.get("/t")

    def test_keep_alive(self, coord):
        coord.put("/t", "v", ttl_ms=300)
        for _ in range(4):
            time.sleep(0.15)
            coord.keep_alive("/t", 300)
        assert coord.get("/t") == "v"

    def test_cas(self, coord):
        assert coord.cas("/c", "", True, "v1") is True
        assert coord.cas("/c", "", True, "v2") is False
        assert coord.cas("/c", "v1", False, "v2") is True
        assert coord.get("/c") == "v2"

    def test_watch_events(self, coord):
        events = []
        wid = coord.watch_prefix("/w/", lambda ev: events.append(ev))
        coord.put("/w/a", "1")
        coord.delete_("/w/a")
        coord.put("/x/other", "ignored")
        time.sleep(0.1)
        assert [e.type for e in events] == [bb.EventType.PUT, bb.EventType.DELETE]
        coord.unwatch(wid)
        coord.put("/w/b", "2")
        time.sleep(0.1)
        assert len(events) == 2

    def test_watch_expire_event(self, coord):
        events = []
        coord.watch_prefix("/hb/", lambda ev: events.append(ev))
        coord.put("/hb/w1", "alive", ttl_ms=100)
        time.sleep(0.6)
        types = [e.type for e in events]
        assert bb.EventType.EXPIRE in types


class TestTcp:
    def test_remote_roundtrip(self, coord_server):
        c = bb.CoordClient()
        c.connect("127.0.0.1:%d" % coord_server.port)
        c.put("/r", "v")
        assert c.get("/r") == "v"
        assert c.get_prefix("/r") == [("/r", "v")]
        c.close()

    def test_remote_watch_and_disconnect_cleanup(self, coord_server):
        c1 = bb.CoordClient()
        c1.connect("127.0.0.1:%d" % coord_server.port)
        events = []
        c1.watch_prefix("/w/", lambda ev: events.append(ev.key))
        c2 = bb.CoordClient()
        c2.connect("127.0.0.1:%d" % coord_server.port)
        c2.put("/w/k", "v")
        time.sleep(0.2)
        assert events == ["/w/k"]
        c1.close()
        time.sleep(0.1)
        c2.put("/w/k2", "v")  # must not crash the server
        assert c2.get("/w/k2") == "v"
        c2.close()

    def test_two_clients_shared_view(self, coord_server):
        c1, c2 = bb.CoordClient(), bb.CoordClient()
        c1.connect("127.0.0.1:%d" % coord_server.port)
        c2.connect("127.0.0.1:%d" % coord_server.port)
        c1.put("/s", "from-c1")
        assert c2.get("/s") == "from-c1"
        c1.close()
        c2.close()


class TestLeaderElection:
    def test_single_candidate_wins(self, coord):
        e = bb.LeaderElector(coord, "/elect", "cand-1", lease_ms=300)
        e.start()
        deadline = time.time() + 2
        while time.time() < deadline and not e.is_leader:
            time.sleep(0.02)
        assert e.is_leader
        assert e.current_leader() == "cand-1"
        e.stop()

    def test_failover(self, coord):
        e1 = bb.LeaderElector(coord, "/elect", "cand-1", lease_ms=300)
        e2 = bb.LeaderElector(coord, "/elect", "cand-2", lease_ms=300)
        e1.start()
        deadline = time.time() + 2
        while time.time() < deadline and not e1.is_leader:
            time.sleep(0.02)
        assert e1.is_leader
        e2.start()
        time.sleep(0.5)
        assert not e2.is_leader  # e1 holds the lease
        e1.stop()  # resigns
        deadline = time.time() + 3
        while time.time() < deadline and not e2.is_leader:
            time.sleep(0.05)
        assert e2.is_leader
        e2.stop()


class TestReconnect:
    def test_client_survives_server_restart(self):
        """CoordClient auto-redials after a coordination outage: calls
        recover and watches fire on the NEW server (state is repopulated by
        the writer, as workers do)."""
        s1 = bb.CoordServer()
        s1.start("127.0.0.1", 0)
        port = s1.port
        c = bb.CoordClient()
        c.connect("127.0.0.1:%d" % port)
        events = []
        c.watch_prefix("/w/", lambda ev: events.append(ev.key))
        c.put("/w/a", "1")
        time.sleep(0.2)
        assert events == ["/w/a"]
        s1.stop()
        time.sleep(0.2)
        s2 = bb.CoordServer()
        s2.start("127.0.0.1", port)  # fresh (empty) server on the same port
        try:
            c.put("/w/b", "2")  # triggers redial + watch re-subscribe
            assert c.get("/w/b") == "2"
            time.sleep(0.3)
            assert "/w/b" in events
        finally:
            c.close()
            s2.stop()

    def test_cluster_survives_coord_restart(self):
        """Full-cluster outage drill: the coordination server restarts empty;
        the worker re-registers via its reconnect hook and the keystone
        rescans — objects stay served throughout (metadata is keystone's)."""
        import os as _os
        from conftest import Cluster
        s1 = bb.CoordServer()
        s1.start("127.0.0.1", 0)
        port = s1.port
        ep = "127.0.0.1:%d" % port
        kc = bb.KeystoneConfig()
        kc.listen_address = "127.0.0.1:0"
        kc.coord_endpoint = ep
        kc.gc_interval_ms = 100000
        srv = bb.create_and_start_keystone(kc)
        wc = bb.WorkerConfig()
        wc.worker_id = "rw0"
        wc.coord_endpoint = ep
        wc.data_listen_address = "127.0.0.1:0"
        wc.heartbeat_interval_ms = 200
        wc.heartbeat_ttl_ms = 2000
        p = bb.PoolConfig()
        p.pool_id = "rpool"
        p.storage_class = bb.StorageClass.RAM_CPU
        p.size_bytes = 64 << 20
        wc.pools = [p]
        w = bb.WorkerService(wc)
        w.initialize()
        w.start()
        deadline = time.time() + 5
        while time.time() < deadline and not srv.service().get_memory_pools():
            time.sleep(0.02)
        o = bb.ClientOptions()
        o.keystone_endpoint = srv.endpoint
        c = bb.Client(o)
        c.connect()
        data = _os.urandom(256 * 1024)
        c.put("durable", data)
        try:
            s1.stop()
            time.sleep(0.5)
            s2 = bb.CoordServer()
            s2.start("127.0.0.1", port)
            # worker heartbeat loop hits the dead conn, redials, re-registers
            deadline = time.time() + 8
            while time.time() < deadline:
                ws = srv.service().get_workers_info()
                if ws and ws[0].worker_id == "rw0":
                    break
                time.sleep(0.1)
            assert srv.service().get_workers_info(), "worker did not re-register"
            # the object survived the outage
            assert c.get("durable") == data
            s2.stop()
        finally:
            c.close()
            w.stop()
            srv.stop()
            srv.service().stop()


class TestDurability:
    def test_snapshot_roundtrip(self, tmp_path):
        """CoordStore save/load: values and TTL deadlines survive; entries
        that expired while the store was down are dropped on load."""
        snap = str(tmp_path / "coord.snap")
        s1 = bb.CoordStore()
        s1.put("/persist/a", "alpha")
        s1.put("/persist/b", "beta")
        s1.put("/persist/ephemeral", "soon-gone", 250)  # ttl_ms
        s1.put("/persist/long", "stays", 60000)
        assert s1.dirty()
        s1.save(snap)
        assert not s1.dirty()
        time.sleep(0.4)  # ephemeral expires while "down"
        s2 = bb.CoordStore()
        s2.load(snap)
        assert s2.get("/persist/a") == "alpha"
        assert s2.get("/persist/b") == "beta"
        assert s2.get("/persist/long") == "stays"
        with pytest.raises(Exception, match="KEY_NOT_FOUND"):
            s2.get("/persist/ephemeral")

    def test_load_missing_and_corrupt(self, tmp_path):
        s = bb.CoordStore()
        with pytest.raises(Exception, match="KEY_NOT_FOUND"):
            s.load(str(tmp_path / "nope.snap"))
        bad = tmp_path / "bad.snap"
        bad.write_bytes(b"not a snapshot")
        with pytest.raises(Exception, match="PROTOCOL_ERROR"):
            s.load(str(bad))

    def test_coordd_daemon_restart_keeps_data(self, tmp_path):
        """coordd --data-dir: a restart of the daemon restores the keyspace
        (the reference's deployments got this from etcd's WAL)."""
        import os
        import signal
        import socket
        import subprocess
        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        coordd = os.path.join(repo, "bin", "coordd")
        if not os.path.exists(coordd):
            pytest.skip("daemons not built")
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()

        def start():
            f = open(tmp_path / "coordd.log", "a")
            return subprocess.Popen(
                [coordd, "--listen-host", "127.0.0.1",
                 "--listen-port", str(port),
                 "--data-dir", str(tmp_path / "data")],
                stdout=f, stderr=subprocess.STDOUT)

        p = start()
        try:
            c = bb.CoordClient()
            deadline = time.time() + 5
            while time.time() < deadline:
                try:
                    c.connect("127.0.0.1:%d" % port)
                    break
                except Exception:
                    time.sleep(0.1)
            c.put("/dur/x", "42")
            c.put("/dur/y", "43")
            # wait for the once-a-second snapshot tick
            deadline = time.time() + 5
            while time.time() < deadline:
                if (tmp_path / "data" / "coord.snap").exists():
                    break
                time.sleep(0.1)
            c.close()
            p.send_signal(signal.SIGTERM)
            p.wait(timeout=5)
            p = start()
            c2 = bb.CoordClient()
            deadline = time.time() + 5
            while time.time() < deadline:
                try:
                    c2.connect("127.0.0.1:%d" % port)
                    break
                except Exception:
                    time.sleep(0.1)
            assert c2.get("/dur/x") == "42"
            assert c2.get("/dur/y") == "43"
            c2.close()
        finally:
            p.send_signal(signal.SIGTERM)
            try:
                p.wait(timeout=5)
            except subprocess.TimeoutExpired:
                p.kill()


class TestReplication:
    def _primary(self):
        a = bb.CoordServer()
        a.start("127.0.0.1", 0)
        return a, "127.0.0.1:%d" % a.port

    def test_follower_mirrors_and_ttl(self):
        a, ep_a = self._primary()
        b = bb.CoordServer()
        b.start("127.0.0.1", 0)
        try:
            ca = bb.CoordClient()
            ca.connect(ep_a)
            ca.put("/r/x", "1")
            ca.put("/r/lease", "alive", 400)  # ttl
            f = bb.CoordFollower(b, ep_a, failover_ms=60000)
            f.start()
            assert b.read_only()
            # bootstrap dump arrived
            assert b.store().get("/r/x") == "1"
            assert b.store().get("/r/lease") == "alive"
            # live event stream mirrors new writes
            ca.put("/r/y", "2")
            deadline = time.time() + 3
            while time.time() < deadline:
                try:
                    if b.store().get("/r/y") == "2":
                        break
                except Exception:
                    pass
                time.sleep(0.02)
            assert b.store().get("/r/y") == "2"
            # delete propagates
            ca.delete_("/r/x")
            deadline = time.time() + 3
            while time.time() < deadline:
                try:
                    b.store().get("/r/x")
                    time.sleep(0.02)
                except Exception:
                    break
            with pytest.raises(Exception):
                b.store().get("/r/x")
            # the TTL was mirrored: the lease expires on the standby too
            time.sleep(0.6)
            b.store().sweep_now()
            with pytest.raises(Exception):
                b.store().get("/r/lease")
            f.stop()
            ca.close()
        finally:
            b.stop()
            a.stop()

    def test_standby_rejects_writes_and_client_cycles(self):
        a, ep_a = self._primary()
        b = bb.CoordServer()
        b.start("127.0.0.1", 0)
        ep_b = "127.0.0.1:%d" % b.port
        try:
            f = bb.CoordFollower(b, ep_a, failover_ms=60000)
            f.start()
            # direct client on the standby: mutation refused
            cb = bb.CoordClient()
            cb.connect(ep_b)
            with pytest.raises(Exception, match="NOT_LEADER"):
                cb.put("/w/k", "v")
            cb.close()
            # a client given "standby,primary" transparently lands the write
            c = bb.CoordClient()
            c.connect("%s,%s" % (ep_b, ep_a))
            c.put("/w/k", "v")
            assert c.get("/w/k") == "v"
            c.close()
            f.stop()
        finally:
            b.stop()
            a.stop()

    def test_failover_promotes_standby(self):
        a, ep_a = self._primary()
        b = bb.CoordServer()
        b.start("127.0.0.1", 0)
        ep_b = "127.0.0.1:%d" % b.port
        c = bb.CoordClient()
        try:
            f = bb.CoordFollower(b, ep_a, failover_ms=500)
            f.start()
            c.connect("%s,%s" % (ep_a, ep_b))
            c.put("/f/k", "before")
            time.sleep(0.2)  # let replication stream it
            a.stop()
            deadline = time.time() + 6
            while time.time() < deadline and not f.promoted():
                time.sleep(0.05)
            assert f.promoted()
            assert not b.read_only()
            # the same client keeps working (cycles to the promoted standby)
            assert c.get("/f/k") == "before"
            c.put("/f/after", "ok")
            assert c.get("/f/after") == "ok"
            f.stop()
            c.close()
        finally:
            b.stop()

    def test_cluster_survives_coord_failover(self):
        """Primary+standby coordd: the whole cluster (keystone, worker,
        client) is configured with "epA,epB"; killing the primary promotes
        the standby and everything keeps serving — no restart, no data loss
        in the coordination state."""
        import os as _os
        a = bb.CoordServer()
        a.start("127.0.0.1", 0)
        ep_a = "127.0.0.1:%d" % a.port
        b = bb.CoordServer()
        b.start("127.0.0.1", 0)
        ep_b = "127.0.0.1:%d" % b.port
        both = "%s,%s" % (ep_a, ep_b)
        f = bb.CoordFollower(b, ep_a, failover_ms=500)
        f.start()
        kc = bb.KeystoneConfig()
        kc.listen_address = "127.0.0.1:0"
        kc.coord_endpoint = both
        kc.gc_interval_ms = 100000
        srv = bb.create_and_start_keystone(kc)
        wc = bb.WorkerConfig()
        wc.worker_id = "fw0"
        wc.coord_endpoint = both
        wc.data_listen_address = "127.0.0.1:0"
        wc.heartbeat_interval_ms = 200
        wc.heartbeat_ttl_ms = 3000
        p = bb.PoolConfig()
        p.pool_id = "fpool"
        p.storage_class = bb.StorageClass.RAM_CPU
        p.size_bytes = 64 << 20
        wc.pools = [p]
        w = bb.WorkerService(wc)
        w.initialize()
        w.start()
        deadline = time.time() + 5
        while time.time() < deadline and not srv.service().get_memory_pools():
            time.sleep(0.02)
        o = bb.ClientOptions()
        o.keystone_endpoint = srv.endpoint
        c = bb.Client(o)
        c.connect()
        data = _os.urandom(256 * 1024)
        c.put("ha-obj", data)
        try:
            time.sleep(0.3)  # replication streams registrations to B
            a.stop()
            deadline = time.time() + 6
            while time.time() < deadline and not f.promoted():
                time.sleep(0.05)
            assert f.promoted()
            # object survives; worker keeps heartbeating against B; new
            # puts work end to end
            assert c.get("ha-obj") == data
            deadline = time.time() + 8
            ok = False
            while time.time() < deadline:
                try:
                    c.put("post-failover", b"fresh")
                    ok = True
                    break
                except Exception:
                    time.sleep(0.2)
            assert ok
            assert c.get("post-failover") == b"fresh"
            # worker still registered (liveness flows through B now)
            time.sleep(1.0)
            assert len(c.workers_info()) == 1
        finally:
            c.close()
            w.stop()
            f.stop()
            srv.stop()
            srv.service().stop()
            b.stop()


class TestPutManyReplication:
    def test_put_many_streams_to_follower(self):
        """The batched durable-ack mutation (PUT_MANY) must replicate to a
        standby exactly like individual puts/dels — the promoted standby
        carries every object the primary acked."""
        a = bb.CoordServer()
        a.start("127.0.0.1", 0)
        ep_a = "127.0.0.1:%d" % a.port
        b = bb.CoordServer()
        b.start("127.0.0.1", 0)
        c = bb.CoordClient()
        try:
            f = bb.CoordFollower(b, ep_a, failover_ms=60000)
            f.start()
            c.connect(ep_a)
            c.put_many([("/pm/a", "1"), ("/pm/b", "2"), ("/pm/c", "3")], [])
            c.put_many([("/pm/d", "4")], ["/pm/b"])
            deadline = time.time() + 5
            while time.time() < deadline:
                try:
                    if (b.store().get("/pm/d") == "4" and
                            b.store().get("/pm/a") == "1"):
                        break
                except Exception:
                    pass
                time.sleep(0.05)
            assert b.store().get("/pm/a") == "1"
            assert b.store().get("/pm/c") == "3"
            assert b.store().get("/pm/d") == "4"
            with pytest.raises(Exception):
                b.store().get("/pm/b")  # deleted in the same batch
            f.stop()
        finally:
            c.close()
            b.stop()
            a.stop()


class TestFencing:
    """Leadership fencing epochs: a deposed primary that
    revives AFTER standby promotion must not accept writes. Clients carry the
    highest epoch they observed; a server seeing a newer epoch than its own
    permanently demotes itself."""

    def test_revived_primary_is_fenced(self):
        a = bb.CoordServer()
        a.start("127.0.0.1", 0)
        ep_a = "127.0.0.1:%d" % a.port
        store_a = a.store()
        b = bb.CoordServer()
        b.start("127.0.0.1", 0)
        ep_b = "127.0.0.1:%d" % b.port
        a2 = None
        c = bb.CoordClient()
        c2 = bb.CoordClient()
        try:
            f = bb.CoordFollower(b, ep_a, failover_ms=400)
            f.start()
            c.connect("%s,%s" % (ep_a, ep_b))
            epoch0 = store_a.epoch()
            c.put("/z/k", "v1")
            time.sleep(0.3)  # replication streams it
            a.stop()
            deadline = time.time() + 6
            while time.time() < deadline and not f.promoted():
                time.sleep(0.05)
            assert f.promoted()
            # promotion bumped the fencing epoch
            assert b.store().epoch() == epoch0 + 1
            c.put("/z/k", "v2")  # client fails over to B, observes new epoch
            assert c.observed_epoch == epoch0 + 1

            # REVIVE the old primary: same store (stale epoch), new listener
            a2 = bb.CoordServer(store_a)
            a2.start("127.0.0.1", 0)
            ep_a2 = "127.0.0.1:%d" % a2.port
            # the failed-over client reaches the revived primary: its write
            # carries the newer epoch and must be REFUSED
            c.connect(ep_a2)
            with pytest.raises(Exception, match="NOT_LEADER"):
                c.put("/z/k", "v3-split-brain")
            # ...and the refusal FENCED the revived primary permanently:
            # even a fresh, epoch-ignorant client is refused now
            assert a2.read_only()
            c2.connect(ep_a2)
            with pytest.raises(Exception, match="NOT_LEADER"):
                c2.put("/z/other", "x")
            # the batched mutation path (durable-ack commits) is fenced too
            with pytest.raises(Exception, match="NOT_LEADER"):
                c2.put_many([("/z/pm", "y")], ["/z/k"])
            # the split-brain write never landed anywhere
            assert b.store().get("/z/k") == "v2"
            with pytest.raises(Exception):
                store_a.get("/z/other")
            f.stop()
        finally:
            c.close()
            c2.close()
            if a2 is not None:
                a2.stop()
            b.stop()
            a.stop()

    def test_follower_refuses_stale_primary(self):
        # a promoted store (higher epoch) must never re-follow an old-epoch
        # primary — mirroring it would resurrect the split brain
        a = bb.CoordServer()
        a.start("127.0.0.1", 0)
        ep_a = "127.0.0.1:%d" % a.port
        b = bb.CoordServer()
        b.start("127.0.0.1", 0)
        try:
            b.store().bump_epoch()  # b was promoted at some point
            f = bb.CoordFollower(b, ep_a, failover_ms=60000)
            with pytest.raises(Exception, match="stale epoch|NOT_LEADER"):
                f.start()
        finally:
            b.stop()
            a.stop()

    def test_epoch_persists_in_snapshot(self, tmp_path):
        s = bb.CoordStore()
        s.put("/p/k", "v")
        s.bump_epoch()
        s.bump_epoch()
        e = s.epoch()
        path = str(tmp_path / "snap.bb")
        s.save(path)
        s2 = bb.CoordStore()
        s2.load(path)
        assert s2.epoch() == e
        assert s2.get("/p/k") == "v"
