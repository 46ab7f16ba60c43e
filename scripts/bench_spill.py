#!/usr/bin/env python3
"""BASELINE config #4 benchmark: GPU→DRAM→NVMe tiered eviction under a
working set 2× the HBM pool. Objects accumulate (no removes); the tier
manager demotes cold objects out of HBM as it fills; gets sample the whole
key space so reads hit every tier. Reports sustained put+get throughput and
the final tier distribution. One GPU (run per-rank for more)."""
import json
import os
import statistics
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import blackbird_amd as bb  # noqa: E402

MB = 1 << 20
HBM_POOL = int(os.environ.get("BB_SPILL_HBM", str(2048 * MB)))
OBJ = int(os.environ.get("BB_SPILL_OBJ", str(1 * MB)))
STEPS = int(os.environ.get("BB_SPILL_STEPS", "40"))
BATCH = int(os.environ.get("BB_SPILL_BATCH", "128"))
# working set after all steps = STEPS × BATCH × OBJ = 2.5× HBM_POOL default

use_gpu = bb.core.gpu.available()

cs = bb.CoordServer(); cs.start("unix:/tmp/bb-spill-coord.sock", 0)
ep = cs.endpoint
kc = bb.KeystoneConfig()
kc.listen_address = "unix:/tmp/bb-spill-ks.sock"
kc.coord_endpoint = ep
kc.gc_interval_ms = 500          # tiering runs continuously
kc.tier_high_watermark = 0.80
kc.promote_hot_threshold = 0
srv = bb.create_and_start_keystone(kc)

tmp = tempfile.mkdtemp(prefix="bb-spill-")
wc = bb.WorkerConfig(); wc.worker_id = "spill0"; wc.coord_endpoint = ep
wc.data_listen_address = "127.0.0.1:0"
pools = []
if use_gpu:
    g = bb.PoolConfig(); g.pool_id = "hbm"; g.storage_class = bb.StorageClass.RAM_GPU
    g.size_bytes = HBM_POOL; pools.append(g)
else:  # CPU fallback: DRAM plays the fast tier
    g = bb.PoolConfig(); g.pool_id = "fast"; g.storage_class = bb.StorageClass.RAM_CPU
    g.size_bytes = HBM_POOL; pools.append(g)
pin = bb.PoolConfig(); pin.pool_id = "pinned"; pin.storage_class = bb.StorageClass.PINNED_CPU
pin.size_bytes = HBM_POOL; pools.append(pin)
nv = bb.PoolConfig(); nv.pool_id = "nvme"; nv.storage_class = bb.StorageClass.NVME
nv.size_bytes = 4 * HBM_POOL; nv.mount_path = tmp; pools.append(nv)
wc.pools = pools
w = bb.WorkerService(wc); w.initialize(); w.start()
while len(srv.service().get_memory_pools()) < len(pools):
    time.sleep(0.05)

o = bb.ClientOptions(); o.keystone_endpoint = kc.listen_address
client = bb.Client(o); client.connect()
cfg = bb.PlacementConfig()
cfg.preferred_class = (bb.StorageClass.RAM_GPU if use_gpu
                       else bb.StorageClass.RAM_CPU)

if use_gpu:
    gcl = bb.GpuClient(client, 0); gcl.init()
    src = bb.core.gpu.malloc(BATCH * OBJ)
    dst = bb.core.gpu.malloc(BATCH * OBJ)
    bb.core.gpu.fill_pattern(src, BATCH * OBJ, seed=99)

import random
rng = random.Random(7)
all_keys = []
t0 = time.perf_counter()
bytes_moved = 0
get_lat = []
phases = os.environ.get("BB_SPILL_PHASES") == "1"
for step in range(STEPS):
    keys = ["sp-%d-%d" % (step, i) for i in range(BATCH)]
    tp = time.perf_counter()
    if use_gpu:
        st = gcl.batch_put_device([(k, src + i * OBJ, OBJ)
                                   for i, k in enumerate(keys)], cfg)
    else:
        st = client.batch_put([(k, b"\xab" * OBJ) for k in keys], cfg)
    put_ms = (time.perf_counter() - tp) * 1e3
    placed = [k for k, s in zip(keys, st) if s == 0]
    all_keys.extend(placed)
    bytes_moved += len(placed) * OBJ
    # read a batch sampled across the WHOLE history (every tier gets hit)
    sample = rng.sample(all_keys, min(BATCH, len(all_keys)))
    tg = time.perf_counter()
    if use_gpu:
        st = gcl.batch_get_device([(k, dst + i * OBJ, OBJ)
                                   for i, k in enumerate(sample)])
    else:
        st = [s for s, _ in client.batch_get(sample)]
    get_lat.append((time.perf_counter() - tg) * 1e3)
    ok = sum(1 for s in st if s == 0)
    assert ok == len(sample), st[:5]
    bytes_moved += ok * OBJ
    if phases:
        print("[phase] step=%d put_ms=%.1f get_ms=%.1f" %
              (step, put_ms, get_lat[-1]), file=sys.stderr, flush=True)
elapsed = time.perf_counter() - t0

dist = {}
ks = srv.service()
for k in all_keys:
    try:
        cls = ks.get_workers(k).copies[0].shards[0].storage_class.name
        dist[cls] = dist.get(cls, 0) + 1
    except Exception:
        dist["LOST"] = dist.get("LOST", 0) + 1

print(json.dumps({
    "metric": "tiered_spill_put_get_throughput",
    "value": round(bytes_moved / elapsed / 1e9, 3),
    "unit": "GB/s",
    "working_set_bytes": len(all_keys) * OBJ,
    "fast_tier_bytes": HBM_POOL,
    "working_set_over_fast_tier": round(len(all_keys) * OBJ / HBM_POOL, 2),
    "p50_get_batch_ms": round(statistics.median(get_lat), 2),
    "tier_distribution": dist,
    "tier": "RAM_GPU" if use_gpu else "RAM_CPU",
    "data": "synthetic",
}))

client.close(); w.stop(); srv.stop(); srv.service().stop(); cs.stop()
