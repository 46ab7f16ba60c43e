#!/usr/bin/env python3
"""Flagship benchmark: batched put/get of 1 MiB objects through the
MI355X-native object store (BASELINE.json metric: "batched put/get throughput
(GB/s whole node) + p50 get latency, 1 MiB objects at 1/2/4/8 workers").

Topology per rank (one rank per GPU, launched by torch.distributed.run):
  rank 0 additionally hosts the coordination server + keystone;
  every rank runs one worker with an HBM pool on its GPU (hipMalloc), and one
  client whose source/destination buffers live in its GPU's HBM.

One step = each rank batch-puts B objects of S bytes from device memory into
the cluster (placement spreads across all workers → (N-1)/N of traffic
crosses xGMI via hipIpc one-sided copies), batch-gets them back into device
memory, and batch-removes them. Object digests are computed on-GPU by the
MFMA checksum kernel as part of every put (it is a feature of the store, so
it is inside the timed region).

Synthetic data (kernel-generated pseudo-random bytes), no datasets needed.
Without a GPU the bench falls back to the DRAM/SHM tier with host buffers so
the harness logic is testable on CPU (the reported tier is in `config`).
"""
import argparse
import json
import os
import statistics
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

import blackbird_amd as bb  # noqa: E402

MB = 1 << 20


def log(msg):
    print(f"[bench r{RANK}] {msg}", file=sys.stderr, flush=True)


RANK = int(os.environ.get("RANK", "0"))
WORLD = int(os.environ.get("WORLD_SIZE", "1"))
LOCAL_RANK = int(os.environ.get("LOCAL_RANK", str(RANK)))


def pick_device():
    """LOCAL_RANK → device; wraps when ranks exceed visible devices (lets a
    multi-process run be validated on a single-GPU box — cross-process IPC
    behaves the same)."""
    n = bb.core.gpu.device_count()
    return LOCAL_RANK % n if n > 0 else 0


DEVICE = pick_device()


def setup_dist():
    if WORLD == 1:
        return None
    import torch.distributed as dist
    # gloo for control-plane rendezvous/barriers; the store's own data plane
    # (hipIpc over xGMI) is what is being measured, not torch collectives.
    dist.init_process_group("gloo", rank=RANK, world_size=WORLD)
    return dist


def barrier(dist):
    if dist is not None:
        dist.barrier()


def max_over_ranks(dist, value):
    if dist is None:
        return value
    import torch
    t = torch.tensor([value], dtype=torch.float64)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--object-size", type=int, default=1 * MB)
    ap.add_argument("--objects", type=int, default=512,
                    help="objects per rank per step")
    ap.add_argument("--replication", type=int, default=1)
    ap.add_argument("--no-placement-cache", action="store_true",
                    help="disable the digest-verified client placement cache")
    ap.add_argument("--no-fused-copy", action="store_true",
                    help="disable the fused scatter/gather kernel (per-shard hipMemcpyAsync only)")
    ap.add_argument("--tier", choices=["auto", "gpu", "cpu"], default="auto")
    ap.add_argument("--latency-probes", type=int, default=64)
    ap.add_argument("--pipeline", type=int, default=2,
                    help="independent batch lanes in flight (overlaps "
                         "control-plane RPCs with GPU transfers)")
    ap.add_argument("--preset", type=int, choices=[1, 2, 3, 5],
                    help="BASELINE.json config presets: 1=TCP-loopback DRAM "
                         "1KiB, 2=HBM batched 1MiB, 3=replication-2 over "
                         "xGMI (launch with torchrun), 5=64KiB small-object "
                         "batches. Config 4 (tiered spill) is "
                         "scripts/bench_spill.py")
    args = ap.parse_args()
    if args.preset == 1:
        args.tier, args.object_size, args.objects = "cpu", 1024, 512
    elif args.preset == 2:
        args.tier, args.object_size, args.objects = "gpu", 1 << 20, 512
    elif args.preset == 3:
        args.tier, args.object_size, args.objects = "gpu", 1 << 20, 256
        args.replication = 2
    elif args.preset == 5:
        args.tier, args.object_size, args.objects = "gpu", 65536, 4096

    n_gpus = WORLD if WORLD > 1 else args.gpus
    use_gpu = args.tier == "gpu" or (args.tier == "auto" and bb.core.gpu.available())
    dist = setup_dist()

    base_port = int(os.environ.get("MASTER_PORT", "29500"))
    host = "127.0.0.1"
    # all ranks share one node → Unix-domain sockets for the metadata plane
    coord_ep = f"unix:/tmp/bb-bench-{base_port}-coord.sock"
    keystone_ep = f"unix:/tmp/bb-bench-{base_port}-ks.sock"

    # ---- control plane (rank 0) ----
    cs = srv = None
    if RANK == 0:
        store = bb.CoordStore()
        cs = bb.CoordServer(store)
        cs.start(coord_ep, 0)
        kc = bb.KeystoneConfig()
        kc.listen_address = keystone_ep
        kc.coord_endpoint = coord_ep
        # benchmark cadence: maintenance passes (GC/tiering/repair scans)
        # run between measurements, not inside them
        kc.gc_interval_ms = 60000
        kc.promote_hot_threshold = 0
        srv = bb.create_and_start_keystone(kc)
    barrier(dist)

    # ---- worker (every rank) ----
    lanes_hint = max(1, args.pipeline) if use_gpu else 1
    pool_bytes = max(
        args.objects * args.object_size * args.replication * (lanes_hint + 2),
        64 * MB)
    wc = bb.WorkerConfig()
    wc.worker_id = f"w{RANK}"
    wc.coord_endpoint = coord_ep
    wc.data_listen_address = f"{host}:0"
    pc = bb.PoolConfig()
    pc.pool_id = f"hbm{RANK}" if use_gpu else f"dram{RANK}"
    pc.storage_class = (bb.StorageClass.RAM_GPU if use_gpu
                        else bb.StorageClass.RAM_CPU)
    pc.size_bytes = pool_bytes
    pc.gpu_device_id = DEVICE
    wc.pools = [pc]
    worker = bb.WorkerService(wc)
    worker.initialize()
    worker.start()
    barrier(dist)

    # ---- client ----
    opts = bb.ClientOptions()
    opts.keystone_endpoint = keystone_ep
    client = bb.Client(opts)
    deadline = time.time() + 30
    while True:
        try:
            client.connect()
            if len(client.memory_pools()) >= n_gpus:
                break
        except Exception:
            pass
        if time.time() > deadline:
            raise TimeoutError("cluster did not assemble")
        time.sleep(0.1)

    cfg = bb.PlacementConfig()
    cfg.replication = args.replication
    cfg.checksum = True
    cfg.preferred_class = (bb.StorageClass.RAM_GPU if use_gpu
                           else bb.StorageClass.RAM_CPU)
    # locality: each rank keeps its working set in its own GPU's HBM
    # (remote xGMI access still covered by replication / failover paths)
    cfg.preferred_worker = f"w{RANK}"

    B, S = args.objects, args.object_size
    lanes = max(1, args.pipeline)

    if use_gpu:
        # Independent lanes: each has its own GpuClient (streams), source and
        # destination buffers, and key space — steps on different lanes
        # overlap (one lane's metadata RPCs run while another lane's GPU
        # transfers execute). All work still happens; nothing is skipped.
        lane_objs = []
        lane_clients = []
        for L in range(lanes):
            lc = bb.Client(opts)   # own metadata connection per lane
            lc.connect()
            lane_clients.append(lc)
            gcl = bb.GpuClient(lc, DEVICE)
            gcl.init()
            gcl.set_fused_copy(not args.no_fused_copy)
            # verified placement cache: gets of our own puts skip the
            # metadata RPC and validate by digest inside the gather kernel
            gcl.set_placement_cache(not args.no_placement_cache)
            src = bb.core.gpu.malloc(B * S, DEVICE)
            dst = bb.core.gpu.malloc(B * S, DEVICE)
            bb.core.gpu.fill_pattern(src, B * S, seed=1234 + RANK * 17 + L)
            pb = bb.make_put_batch(
                [(f"r{RANK}L{L}o{i}", src + i * S, S) for i in range(B)])
            gb = bb.make_get_batch(
                [(f"r{RANK}L{L}o{i}", dst + i * S, S) for i in range(B)])
            lane_objs.append((gcl, src, dst, pb, gb))
        gcl, src, dst, put_batch, get_batch = lane_objs[0]

        phase_log = os.environ.get("BB_BENCH_PHASES") == "1"

        # upsert puts: each step's put atomically replaces last step's
        # object under the same key — no separate remove RPC in the loop
        # (BB_BENCH_NO_REPLACE=1 falls back to explicit removes)
        no_replace = os.environ.get("BB_BENCH_NO_REPLACE") == "1"
        cfg.replace = not no_replace

        def do_step(lane=0):
            li = lane % lanes
            g2, _, _, pb, gb = lane_objs[li]
            tp = time.perf_counter()
            assert g2.batch_put_prepared(pb, cfg), "put failures"
            t0 = time.perf_counter()
            assert g2.batch_get_prepared(gb), "get failures"
            t1 = time.perf_counter()
            get_ms = (t1 - t0) * 1e3
            if no_replace:
                assert bb.client_batch_remove_prepared(lane_clients[li], pb)
            if phase_log:
                log(f"phases lane={li} put={1e3*(t0-tp):.2f} get={get_ms:.2f} "
                    f"total={1e3*(t1-tp):.2f}")
            return get_ms
    else:
        import numpy as np
        rng = np.random.default_rng(1234 + RANK)
        # stable buffers (numpy arrays) so the host batch session binds them
        arrs = [rng.integers(0, 256, size=S, dtype=np.uint8) for _ in range(B)]
        put_items = [(f"r{RANK}o{i}", arrs[i]) for i in range(B)]
        keys = [k for k, _ in put_items]
        no_replace = os.environ.get("BB_BENCH_NO_REPLACE") == "1"
        cfg.replace = not no_replace
        host_sess = bb.HostPutSession()

        phase_log = os.environ.get("BB_BENCH_PHASES") == "1"

        def do_step(lane=0):
            tp = time.perf_counter()
            st = client.batch_put_session(put_items, cfg, host_sess)
            assert all(s == 0 for s in st), f"put failures: {st[:5]}"
            t0 = time.perf_counter()
            res = client.batch_get(keys)
            get_ms = (time.perf_counter() - t0) * 1e3
            assert all(s == 0 for s, _ in res)
            if no_replace:
                client.batch_remove(keys)
            if phase_log:
                log(f"phases put={1e3*(t0-tp):.2f} get={get_ms:.2f} "
                    f"fast={client.host_session_steps}")
            return get_ms
        lanes = 1  # host tier: single lane

    def device_sync():
        if use_gpu:
            bb.core.gpu.sync()

    import threading

    def run_steps(n):
        # one thread per lane; each lane executes its own steps strictly
        # sequentially (a lane's key space must never be reused while a
        # previous step on that lane is still in flight)
        if lanes == 1:
            return [do_step(i) for i in range(n)]
        ms_per_lane = [[] for _ in range(lanes)]
        errs = []

        def lane_loop(L):
            try:
                for i in range(L, n, lanes):
                    ms_per_lane[L].append(do_step(L))
            except Exception as e:  # surface assertion details
                errs.append(e)

        ts = [threading.Thread(target=lane_loop, args=(L,)) for L in range(lanes)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        if errs:
            raise errs[0]
        return [m for lane in ms_per_lane for m in lane]

    # ---- warmup ----
    run_steps(args.warmup)
    device_sync()
    barrier(dist)

    # ---- timed region ----
    # CPython's cyclic GC pauses (tens of ms when it strikes) are harness
    # noise, not store behavior — quiesce it for the measured window
    import gc as _gc
    _gc.collect()
    _gc.disable()
    device_sync()
    barrier(dist)
    t0 = time.perf_counter()
    get_batch_ms = run_steps(args.steps)
    device_sync()
    barrier(dist)
    elapsed = time.perf_counter() - t0
    elapsed = max_over_ranks(dist, elapsed)
    _gc.enable()

    # ---- cold regime: same workload with the placement
    # cache OFF and in-place upserts OFF — every step pays the full control
    # plane (batch_put_start allocation + batch_get_workers RPC + explicit
    # removes). Reported as value_cold so regressions on the uncached path
    # are visible in the official record. ----
    cold_gbps = None
    # a regime SAMPLE, not a soak: cap it so huge --steps soaks don't spend
    # longer in the cold phase than the timed region
    cold_steps = max(2, min(args.steps // 4, 200))
    if not use_gpu:
        # host-tier cold regime: fresh keys, full control plane, removes
        cold_cfg = bb.PlacementConfig()
        for attr in ("replication", "checksum", "preferred_class",
                     "preferred_worker"):
            setattr(cold_cfg, attr, getattr(cfg, attr))
        cold_cfg.replace = False
        citems = [(f"r{RANK}cold{i}", arrs[i]) for i in range(B)]
        ckeys = [k for k, _ in citems]

        def cold_step():
            st = client.batch_put(citems, cold_cfg)
            assert all(s == 0 for s in st), "cold put"
            res = client.batch_get(ckeys)
            assert all(s == 0 for s, _ in res)
            client.batch_remove(ckeys)

        cold_step()  # warmup
        barrier(dist)
        t0c = time.perf_counter()
        for _ in range(cold_steps):
            cold_step()
        barrier(dist)
        elapsed_c = max_over_ranks(dist, time.perf_counter() - t0c)
        cold_gbps = 2.0 * B * S * cold_steps * max(WORLD, 1) / elapsed_c / 1e9
    if use_gpu:
        cold_cfg = bb.PlacementConfig()
        for attr in ("replication", "checksum", "preferred_class",
                     "preferred_worker"):
            setattr(cold_cfg, attr, getattr(cfg, attr))
        cold_cfg.replace = False
        gcl_cold = bb.GpuClient(lane_clients[0], DEVICE)
        gcl_cold.init()
        gcl_cold.set_fused_copy(not args.no_fused_copy)
        gcl_cold.set_placement_cache(False)
        ck = [(f"r{RANK}cold{i}", src + i * S, S) for i in range(B)]
        cpb = bb.make_put_batch(ck)
        cgb = bb.make_get_batch([(k, dst + i * S, S)
                                 for i, (k, _, _) in enumerate(ck)])
        cold_keys = [k for k, _, _ in ck]

        def cold_step():
            t0 = time.perf_counter()
            assert gcl_cold.batch_put_prepared(cpb, cold_cfg), "cold put"
            t1 = time.perf_counter()
            assert gcl_cold.batch_get_prepared(cgb), "cold get"
            t2 = time.perf_counter()
            assert bb.client_batch_remove_prepared(lane_clients[0], cpb)
            if phase_log:
                t3 = time.perf_counter()
                log(f"cold phases put={1e3*(t1-t0):.2f} get={1e3*(t2-t1):.2f} "
                    f"rm={1e3*(t3-t2):.2f}")

        cold_step()  # warmup
        device_sync()
        barrier(dist)
        t0c = time.perf_counter()
        for _ in range(cold_steps):
            cold_step()
        device_sync()
        barrier(dist)
        elapsed_c = max_over_ranks(dist, time.perf_counter() - t0c)
        cold_gbps = 2.0 * B * S * cold_steps * max(WORLD, 1) / elapsed_c / 1e9

    # ---- multi-rank shuffle A/B: the symmetric "every rank
    # fetches a peer's batch" pattern via (a) the one-sided IPC gather path
    # and (b) the RCCL all-to-all batch shuffle. Untimed side-measurement;
    # both numbers go into config so the driver's 2/4/8-GPU runs produce the
    # comparison. ----
    def all_ranks_ok(flag):
        """Consensus gate: a rank that failed setup must not leave peers
        blocked at a barrier — every timed phase starts only if ALL agree."""
        if dist is None:
            return flag
        import torch
        t = torch.tensor([1.0 if flag else 0.0], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MIN)
        return t.item() > 0.5

    shuffle_ab = None
    if use_gpu and WORLD > 1:
        peer = (RANK + 1) % WORLD
        peer_keys = [f"r{peer}L0o{i}" for i in range(B)]
        ab_steps = 5
        recv = bb.core.gpu.malloc(B * S, DEVICE)
        shuffle_ab = {}

        # (a) one-sided IPC gather of the peer's batch (cross-GPU xGMI)
        ipc_items = [(k, recv + i * S, S) for i, k in enumerate(peer_keys)]
        gcl_ab = None
        ok = True
        try:
            gcl_ab = bb.GpuClient(lane_clients[0], DEVICE)
            gcl_ab.init()
            gcl_ab.set_fused_copy(not args.no_fused_copy)
            assert gcl_ab.batch_get_device(ipc_items) == [0] * B
            bb.core.gpu.sync()
        except Exception as e:
            log(f"shuffle A/B ipc setup failed: {e}")
            shuffle_ab["ipc_error"] = str(e)[:200]
            ok = False
        if all_ranks_ok(ok):
            barrier(dist)
            t0s = time.perf_counter()
            for _ in range(ab_steps):
                assert gcl_ab.batch_get_device(ipc_items) == [0] * B
            bb.core.gpu.sync()
            barrier(dist)
            ipc_s = max_over_ranks(dist, time.perf_counter() - t0s)
            shuffle_ab["ipc_gather_gbps"] = round(
                B * S * ab_steps * WORLD / ipc_s / 1e9, 1)

        # (b) RCCL all-to-all shuffle of the same batches.
        # ncclCommInitRank blocks until EVERY rank calls it, so the ranks
        # agree they are ready BEFORE anyone enters it — a rank failing
        # during setup must not leave its peers blocked inside RCCL.
        eng = coord_cl = None
        ok = True
        try:
            coord_cl = bb.CoordClient()
            coord_cl.connect(coord_ep)
            eng = bb.RcclEngine()
        except Exception as e:
            log(f"shuffle A/B rccl setup failed: {e}")
            shuffle_ab["rccl_error"] = str(e)[:200]
            ok = False
        if all_ranks_ok(ok):
            try:
                eng.init(coord_cl, "default", f"bench-ab-{base_port}", RANK,
                         WORLD, DEVICE)
            except Exception as e:  # e.g. ranks sharing one device
                log(f"shuffle A/B rccl init failed: {e}")
                shuffle_ab["rccl_error"] = str(e)[:200]
                ok = False
        if all_ranks_ok(ok):  # collective: every rank must participate
            wants = [([], [], 0)] * WORLD
            wants[peer] = (peer_keys, [S] * B, recv)
            bb.core.gpu_batch_shuffle(gcl, eng, wants)  # warmup
            bb.core.gpu.sync()
            barrier(dist)
            t0s = time.perf_counter()
            for _ in range(ab_steps):
                bb.core.gpu_batch_shuffle(gcl, eng, wants)
            bb.core.gpu.sync()
            barrier(dist)
            rccl_s = max_over_ranks(dist, time.perf_counter() - t0s)
            shuffle_ab["rccl_alltoall_gbps"] = round(
                B * S * ab_steps * WORLD / rccl_s / 1e9, 1)
            log(f"shuffle A/B: {shuffle_ab}")
        if eng is not None:
            eng.destroy()
        if coord_cl is not None:
            coord_cl.close()
        bb.core.gpu.free(recv)

    # ---- p50 single-object get latency (untimed probe, after the region) ----
    probe_us = []
    probe_key = f"r{RANK}probe"
    if use_gpu:
        gcl.put_device(probe_key, src, S, cfg)
        for _ in range(args.latency_probes):
            t = time.perf_counter()
            gcl.get_device(probe_key, dst, S)
            probe_us.append((time.perf_counter() - t) * 1e6)
    else:
        client.put(probe_key, put_items[0][1], cfg)
        for _ in range(args.latency_probes):
            t = time.perf_counter()
            client.get(probe_key)
            probe_us.append((time.perf_counter() - t) * 1e6)
    client.remove(probe_key)
    p50_us = statistics.median(probe_us)

    # whole-job application bytes: every rank puts B*S and gets B*S per step
    total_bytes = 2.0 * B * S * args.steps * max(WORLD, 1)
    gbps = total_bytes / elapsed / 1e9
    ms_per_step = elapsed / args.steps * 1e3

    if RANK == 0:
        out = {
            "metric": "batched_put_get_throughput",
            "value": round(gbps, 3),
            "unit": "GB/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            # same workload, placement cache off + upserts off: every step
            # pays full allocation + metadata RPCs + removes
            "value_cold": round(cold_gbps, 3) if cold_gbps else None,
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": "uint8",
            "data": "synthetic",
            "config": {
                "model": "tiered-object-store",
                "object_size": S,
                "objects_per_rank_per_step": B,
                "global_batch": B * max(WORLD, 1),
                "seq_len": None,
                "replication": args.replication,
                "tier": "RAM_GPU" if use_gpu else "RAM_CPU",
                "parallelism": f"{n_gpus} workers, 1 HBM pool/GPU, xGMI IPC",
                "checksum": "mfma-bbhash64" if use_gpu else "cpu-bbhash64",
                "fused_copy": not args.no_fused_copy,
                "pipeline_lanes": lanes,
                "p50_get_latency_us": round(p50_us, 1),
                "shuffle_ab": shuffle_ab,
            },
        }
        print(json.dumps(out), flush=True)

    # ---- teardown ----
    barrier(dist)
    if use_gpu:
        for _, s_, d_, _, _ in lane_objs:
            bb.core.gpu.free(s_)
            bb.core.gpu.free(d_)
        for lc in lane_clients:
            lc.close()
    client.close()
    worker.stop()
    if RANK == 0:
        srv.stop()
        srv.service().stop()
        cs.stop()
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
