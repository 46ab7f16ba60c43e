#!/usr/bin/env python3
"""HBM capacity soak: fill a large fraction of one MI355X's 288 GB HBM with
64 MiB objects through the fused put path (digest on every put), then read a
random sample back with digest verification. Validates that pools, slab
allocation, placement and the one-sided paths behave at hundreds of GB, not
just benchmark-sized working sets."""
import json
import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import blackbird_amd as bb  # noqa: E402

GB = 1 << 30
MB = 1 << 20
POOL = int(os.environ.get("BB_CAP_POOL_GB", "200")) * GB
OBJ = int(os.environ.get("BB_CAP_OBJ_MB", "64")) * MB
TARGET = int(os.environ.get("BB_CAP_TARGET_GB", "128")) * GB
BATCH = int(os.environ.get("BB_CAP_BATCH", "16"))  # objects per put batch
SAMPLES = int(os.environ.get("BB_CAP_SAMPLES", "64"))

assert bb.core.gpu.available(), "needs an MI355X"
g = bb.core.gpu

cs = bb.CoordServer(); cs.start("unix:/tmp/bb-cap-coord.sock", 0)
kc = bb.KeystoneConfig()
kc.listen_address = "unix:/tmp/bb-cap-ks.sock"
kc.coord_endpoint = cs.endpoint
kc.gc_interval_ms = 60000
srv = bb.create_and_start_keystone(kc)
wc = bb.WorkerConfig(); wc.worker_id = "cap0"; wc.coord_endpoint = cs.endpoint
wc.data_listen_address = "127.0.0.1:0"
pc = bb.PoolConfig(); pc.pool_id = "hbm-cap"
pc.storage_class = bb.StorageClass.RAM_GPU
pc.size_bytes = POOL
pc.gpu_device_id = 0
wc.pools = [pc]
w = bb.WorkerService(wc); w.initialize(); w.start()
while not srv.service().get_memory_pools():
    time.sleep(0.05)

o = bb.ClientOptions(); o.keystone_endpoint = kc.listen_address
c = bb.Client(o); c.connect()
gcl = bb.GpuClient(c, 0); gcl.init()
cfg = bb.PlacementConfig(); cfg.checksum = True
src = g.malloc(BATCH * OBJ)
dst = g.malloc(OBJ)

n_objs = TARGET // OBJ
keys = []
t0 = time.perf_counter()
i = 0
while i < n_objs:
    nb = min(BATCH, n_objs - i)
    bkeys = ["cap-%06d" % (i + j) for j in range(nb)]
    for j in range(nb):  # distinct contents per object
        g.fill_pattern(src + j * OBJ, OBJ, seed=1000 + i + j)
    g.sync()
    st = gcl.batch_put_device(
        [(k, src + j * OBJ, OBJ) for j, k in enumerate(bkeys)], cfg)
    assert all(s == 0 for s in st), (i, st[:4])
    keys.extend(bkeys)
    i += nb
put_s = time.perf_counter() - t0
stored = n_objs * OBJ

stats = c.cluster_stats()
rng = random.Random(11)
sample = rng.sample(range(len(keys)), min(SAMPLES, len(keys)))
t0 = time.perf_counter()
for idx in sample:
    st = gcl.batch_get_device([(keys[idx], dst, OBJ)], verify=True)
    assert st == [0], (idx, st)
    g.sync()
    assert g.verify_pattern(dst, OBJ, seed=1000 + idx) == 0, idx
get_s = time.perf_counter() - t0

print(json.dumps({
    "metric": "hbm_capacity_soak",
    "stored_bytes": stored,
    "pool_bytes": POOL,
    "objects": n_objs,
    "object_size": OBJ,
    "put_gbps": round(stored / put_s / 1e9, 1),
    "sampled_verified_get_gbps": round(len(sample) * OBJ / get_s / 1e9, 1),
    "cluster_used_bytes": stats.total_used,
    "data": "synthetic",
}), flush=True)
c.close(); w.stop(); srv.stop(); srv.service().stop(); cs.stop()
