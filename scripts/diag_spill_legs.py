#!/usr/bin/env python3
"""Isolate the per-tier legs of the spill path (diagnostic, 1 GPU):
batch_put/batch_get of B 1 MiB objects pinned to each storage class, timed
separately, plus a raw backend read/write microbench of the NVMe pool."""
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import blackbird_amd as bb  # noqa: E402

MB = 1 << 20
B = int(os.environ.get("BB_DIAG_BATCH", "128"))
OBJ = int(os.environ.get("BB_DIAG_OBJ", str(1 * MB)))

cs = bb.CoordServer(); cs.start("unix:/tmp/bb-diag-coord.sock", 0)
ep = cs.endpoint
kc = bb.KeystoneConfig()
kc.listen_address = "unix:/tmp/bb-diag-ks.sock"
kc.coord_endpoint = ep
kc.gc_interval_ms = 60000          # no background tiering during the test
kc.promote_hot_threshold = 0
srv = bb.create_and_start_keystone(kc)

tmp = tempfile.mkdtemp(prefix="bb-diag-")
wc = bb.WorkerConfig(); wc.worker_id = "diag0"; wc.coord_endpoint = ep
wc.data_listen_address = "127.0.0.1:0"
pools = []
for pid, cls, size in [
        ("hbm", bb.StorageClass.RAM_GPU, 2048 * MB),
        ("pinned", bb.StorageClass.PINNED_CPU, 2048 * MB),
        ("nvme", bb.StorageClass.NVME, 4096 * MB)]:
    p = bb.PoolConfig(); p.pool_id = pid; p.storage_class = cls
    p.size_bytes = size
    if cls == bb.StorageClass.NVME:
        p.mount_path = tmp
    pools.append(p)
wc.pools = pools
w = bb.WorkerService(wc); w.initialize(); w.start()
while len(srv.service().get_memory_pools()) < len(pools):
    time.sleep(0.05)

o = bb.ClientOptions(); o.keystone_endpoint = kc.listen_address
client = bb.Client(o); client.connect()
gcl = bb.GpuClient(client, 0); gcl.init()
src = bb.core.gpu.malloc(B * OBJ)
dst = bb.core.gpu.malloc(B * OBJ)
bb.core.gpu.fill_pattern(src, B * OBJ, seed=5)

for cls in [bb.StorageClass.RAM_GPU, bb.StorageClass.PINNED_CPU,
            bb.StorageClass.NVME]:
    cfg = bb.PlacementConfig(); cfg.preferred_class = cls
    keys = ["d-%s-%d" % (cls.name, i) for i in range(B)]
    items_p = [(k, src + i * OBJ, OBJ) for i, k in enumerate(keys)]
    items_g = [(k, dst + i * OBJ, OBJ) for i, k in enumerate(keys)]
    t0 = time.perf_counter()
    st = gcl.batch_put_device(items_p, cfg)
    put_s = time.perf_counter() - t0
    assert all(s == 0 for s in st), (cls, st[:5])
    placed = srv.service().get_workers(keys[0]).copies[0].shards[0]
    t0 = time.perf_counter()
    st = gcl.batch_get_device(items_g)
    get_s = time.perf_counter() - t0
    assert all(s == 0 for s in st), (cls, st[:5])
    gb = B * OBJ / 1e9
    print("%-11s placed=%-11s put %6.1f ms (%5.1f GB/s)   get %6.1f ms "
          "(%5.1f GB/s)" % (cls.name, placed.storage_class.name,
                            put_s * 1e3, gb / put_s, get_s * 1e3, gb / get_s),
          flush=True)
    client.batch_remove(keys)

client.close(); w.stop(); srv.stop(); srv.service().stop(); cs.stop()
