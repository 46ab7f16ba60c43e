"""RPC layer edge cases: unknown methods, handler exceptions, oversized and
concurrent traffic, connection lifecycle. (The layer is otherwise covered
indirectly through every coordination/keystone test.)"""
import threading

import pytest

import blackbird_amd as bb


@pytest.fixture
def coord_pair(coord_server):
    c = bb.CoordClient()
    c.connect("127.0.0.1:%d" % coord_server.port)
    yield coord_server, c
    c.close()


class TestRpcEdgeCases:
    def test_large_values_roundtrip(self, coord_pair):
        _, c = coord_pair
        big = "x" * (8 << 20)  # 8 MiB value through the framed protocol
        c.put("/big", big)
        assert c.get("/big") == big

    def test_binary_safe_values(self, coord_pair):
        _, c = coord_pair
        blob = bytes(range(256)).decode("latin1") * 17
        c.put("/bin", blob)
        assert c.get("/bin") == blob

    def test_concurrent_calls_one_connection(self, coord_pair):
        _, c = coord_pair
        errors = []

        def worker(tid):
            try:
                for i in range(100):
                    c.put(f"/t{tid}/k{i}", str(i))
                    assert c.get(f"/t{tid}/k{i}") == str(i)
            except Exception as e:  # pragma: no cover
                errors.append(e)

        ts = [threading.Thread(target=worker, args=(t,)) for t in range(8)]
        [t.start() for t in ts]
        [t.join() for t in ts]
        assert not errors

    def test_server_survives_abrupt_disconnects(self, coord_server):
        import socket
        for _ in range(5):
            s = socket.socket()
            s.connect(("127.0.0.1", coord_server.port))
            s.send(b"\xff" * 10)  # garbage partial frame
            s.close()
        c = bb.CoordClient()
        c.connect("127.0.0.1:%d" % coord_server.port)
        c.put("/alive", "yes")
        assert c.get("/alive") == "yes"
        c.close()

    def test_call_on_closed_client_errors_cleanly(self, coord_server):
        c = bb.CoordClient()
        c.connect("127.0.0.1:%d" % coord_server.port)
        c.close()
        # endpoint still up → auto-redial succeeds and the call completes
        c2 = bb.CoordClient()
        c2.connect("127.0.0.1:%d" % coord_server.port)
        c2.close()
        coord_server.stop()
        with pytest.raises(Exception):
            c2.put("/x", "y")  # server gone → clean error, no hang


class TestHostileFrames:
    """Protocol robustness: raw garbage on the wire must never take the
    server down or wedge other connections."""

    def _sock(self, port):
        import socket
        s = socket.socket()
        s.settimeout(3)
        s.connect(("127.0.0.1", port))
        return s

    def _assert_alive(self, coord_server):
        c = bb.CoordClient()
        c.connect("127.0.0.1:%d" % coord_server.port)
        c.put("/alive", "yes")
        assert c.get("/alive") == "yes"
        c.close()

    def test_random_garbage(self, coord_server):
        import os as _os
        for _ in range(8):
            s = self._sock(coord_server.port)
            s.sendall(_os.urandom(64))
            s.close()
        self._assert_alive(coord_server)

    def test_oversized_length_header(self, coord_server):
        import struct
        s = self._sock(coord_server.port)
        # header: u32 len, u8 kind, pad, u64 id, u16 method (packed layout
        # irrelevant — a huge len must be rejected before any allocation)
        s.sendall(struct.pack("<I", 0xFFFFFFFF) + b"\x00" * 16)
        s.close()
        self._assert_alive(coord_server)

    def test_truncated_frame_then_disconnect(self, coord_server):
        import struct
        s = self._sock(coord_server.port)
        s.sendall(struct.pack("<I", 1024) + b"\x00" * 12)  # promises 1 KiB
        s.close()                                          # never sends it
        self._assert_alive(coord_server)

    def test_unknown_method_is_an_error_not_a_crash(self, coord_pair):
        server, c = coord_pair
        # CoordClient has no raw-call surface; garbage METHOD bytes reach the
        # server through a hand-built frame instead
        import struct
        s = self._sock(server.port)
        # replicate the wire header shape: u32 len, u8 kind(REQUEST=0),
        # 3 pad, u64 id, u16 method, 6 pad — conservative: send a plausible
        # 24-byte header + tiny body; whatever the exact packing, the server
        # must at worst drop the connection
        s.sendall(struct.pack("<IB3xQH6x", 4, 0, 1, 9999) + b"\x00\x01\x02\x03")
        try:
            s.recv(64)
        except Exception:
            pass
        s.close()
        self._assert_alive(server)


class TestDataPlaneHostile:
    """Malformed/hostile requests against the worker's network-exposed data
    port must be rejected with a clean error before any large allocation or
    out-of-bounds memory access (ADVICE round-1 highs)."""

    HDR = "<IBQH"  # u32 body_len, u8 kind, u64 id, u16 method (packed, 15 B)
    DATA_READ = 201
    DATA_PULL = 206

    def _call(self, endpoint, method, body, timeout=10.0):
        import socket
        import struct
        host, port = endpoint.rsplit(":", 1)
        s = socket.create_connection((host, int(port)), timeout=timeout)
        s.settimeout(timeout)
        try:
            s.sendall(struct.pack(self.HDR, len(body), 0, 1, method) + body)
            hdr = b""
            while len(hdr) < 15:
                chunk = s.recv(15 - len(hdr))
                if not chunk:
                    raise ConnectionError("server closed")
                hdr += chunk
            body_len, kind, rid, m = struct.unpack(self.HDR, hdr)
            resp = b""
            while len(resp) < body_len:
                chunk = s.recv(min(65536, body_len - len(resp)))
                if not chunk:
                    raise ConnectionError("short response")
                resp += chunk
            status = struct.unpack("<i", resp[:4])[0]
            (msg_len,) = struct.unpack("<I", resp[4:8])
            msg = resp[8:8 + msg_len].decode(errors="replace")
            return status, msg, resp[8 + msg_len:]
        finally:
            s.close()

    @staticmethod
    def _s(txt):
        import struct
        b = txt.encode()
        return struct.pack("<I", len(b)) + b

    def test_data_read_huge_length_rejected(self, cluster):
        import struct
        pool = cluster.workers[0].pool_descriptors()[0]
        # length = 16 GiB: must be rejected before the response buffer resize
        body = self._s(pool.pool_id) + struct.pack("<QQ", 0, 16 << 30)
        status, msg, _ = self._call(cluster.workers[0].data_endpoint,
                                    self.DATA_READ, body)
        assert status != 0
        assert "out of bounds" in msg

    def test_data_read_beyond_capacity_rejected(self, cluster):
        import struct
        pool = cluster.workers[0].pool_descriptors()[0]
        body = self._s(pool.pool_id) + struct.pack("<QQ", 0, pool.size + 4096)
        status, msg, _ = self._call(cluster.workers[0].data_endpoint,
                                    self.DATA_READ, body)
        assert status != 0

    def test_data_pull_dst_out_of_bounds_rejected(self, cluster):
        import struct
        w = cluster.workers[0]
        pool = w.pool_descriptors()[0]
        # craft a PullReq whose dst range overflows the destination pool and
        # whose src resolves locally (same pool) — without the bounds check
        # this memcpy'd past the pool
        length = 1 << 20
        dst_offset = pool.size - 4096  # dst_offset + length >> capacity
        body = bb.core.encode_pull_req_for_test(
            pool.pool_id, dst_offset, length, pool.pool_id, 0, length)
        status, msg, _ = self._call(w.data_endpoint, self.DATA_PULL, body)
        assert status != 0
        assert "out of bounds" in msg

    def test_data_pull_src_out_of_bounds_rejected(self, cluster):
        w = cluster.workers[0]
        pool = w.pool_descriptors()[0]
        # src slice overruns the (locally resolved) source pool
        length = 1 << 20
        body = bb.core.encode_pull_req_for_test(
            pool.pool_id, 0, length, pool.pool_id, pool.size - 4096, length)
        status, msg, _ = self._call(w.data_endpoint, self.DATA_PULL, body)
        assert status != 0
        assert "out of bounds" in msg

    def test_data_pull_hostile_vector_count_is_protocol_error(self, cluster):
        import struct
        w = cluster.workers[0]
        pool = w.pool_descriptors()[0]
        # PullReq: dst_pool, dst_offset, total_len, srcs — claim 2^32-1 shard
        # placements with no bytes behind them. The serde guard must surface a
        # decode failure (PROTOCOL_ERROR), not a valid empty vector.
        body = (self._s(pool.pool_id) + struct.pack("<QQ", 0, 0)
                + struct.pack("<I", 0xFFFFFFFF))
        status, msg, _ = self._call(w.data_endpoint, self.DATA_PULL, body)
        assert status != 0
        assert "bad request" in msg
