import os, sys, time
import numpy as np
sys.path.insert(0,'/root/repo')
import blackbird_amd as bb
cs = bb.CoordServer(); cs.start("unix:/tmp/bb-hs-coord.sock", 0)
kc = bb.KeystoneConfig(); kc.listen_address="unix:/tmp/bb-hs-ks.sock"; kc.coord_endpoint=cs.endpoint; kc.gc_interval_ms=60000
srv = bb.create_and_start_keystone(kc)
wc = bb.WorkerConfig(); wc.worker_id="w0"; wc.coord_endpoint=cs.endpoint; wc.data_listen_address="127.0.0.1:0"
pc = bb.PoolConfig(); pc.pool_id="dram0"; pc.storage_class=bb.StorageClass.RAM_CPU; pc.size_bytes=256<<20
wc.pools=[pc]; w=bb.WorkerService(wc); w.initialize(); w.start()
while not srv.service().get_memory_pools(): time.sleep(0.02)
o = bb.ClientOptions(); o.keystone_endpoint=kc.listen_address
c = bb.Client(o); c.connect()
B,S = 512,1024
arrs=[np.random.randint(0,256,size=S,dtype=np.uint8) for _ in range(B)]
items=[("k%d"%i, arrs[i]) for i in range(B)]
keys=[k for k,_ in items]
cfg=bb.PlacementConfig(); cfg.replace=True; cfg.checksum=True
sess=bb.HostPutSession()
for it in range(50):
    t0=time.perf_counter(); st=c.batch_put_session(items,cfg,sess); t1=time.perf_counter()
    assert all(s==0 for s in st)
    res=c.batch_get(keys); t2=time.perf_counter()
    globals().setdefault('sp',[]).append((t1-t0)*1e3); globals().setdefault('sg',[]).append((t2-t1)*1e3)
# full-path comparison (fresh keys, no session)
cfg2=bb.PlacementConfig(); cfg2.checksum=True
for it in range(50):
    fit=[("f%d-%d"%(it,i), arrs[i]) for i in range(B)]
    t0=time.perf_counter(); st=c.batch_put(fit,cfg2); t1=time.perf_counter()
    res=c.batch_get([k for k,_ in fit]); t2=time.perf_counter()
    c.batch_remove([k for k,_ in fit]); t3=time.perf_counter()
    globals().setdefault('fp',[]).append((t1-t0)*1e3); globals().setdefault('fg',[]).append((t2-t1)*1e3)
import statistics as st
def rep(tag, v): print(f"{tag}: avg={sum(v)/len(v):.2f} p50={st.median(v):.2f} max={max(v):.2f} over_5ms={sum(1 for x in v if x>5)}", flush=True)
rep("sess_put", sp); rep("sess_get", sg); rep("full_put", fp); rep("full_get", fg)
c.close(); w.stop(); srv.stop(); srv.service().stop(); cs.stop()
os._exit(0)
