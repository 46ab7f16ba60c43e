import json, os, signal, socket, subprocess, sys, time
sys.path.insert(0, '/root/repo')
import blackbird_amd as bb

repo = '/root/repo'
g = bb.core.gpu
MB = 1 << 20

def free_port():
    s = socket.socket(); s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]; s.close(); return p

cp, kp = free_port(), free_port()
procs = []
def spawn(args, log):
    f = open('/tmp/' + log, 'w')
    procs.append(subprocess.Popen(args, stdout=f, stderr=subprocess.STDOUT))

spawn([f"{repo}/bin/coordd", "--listen-host", "127.0.0.1", "--listen-port", str(cp)], "c.log")
time.sleep(0.3)
spawn([f"{repo}/bin/keystoned", "--listen-address", f"127.0.0.1:{kp}",
       "--coord-endpoint", f"127.0.0.1:{cp}", "--metrics-address", "127.0.0.1:0"], "k.log")
time.sleep(0.3)
cfg = {"worker_id": "shmw0", "coord_endpoint": f"127.0.0.1:{cp}",
       "data_listen_address": "127.0.0.1:0",
       "pools": [{"pool_id": "shm-pool", "storage_class": "RAM_CPU",
                  "size_bytes": 512 << 20}]}
open('/tmp/w.json', 'w').write(json.dumps(cfg))
spawn([f"{repo}/bin/workerd", "--config", "/tmp/w.json"], "w.log")

o = bb.ClientOptions(); o.keystone_endpoint = f"127.0.0.1:{kp}"
c = bb.Client(o)
deadline = time.time() + 15
while True:
    try:
        c.connect()
        if c.memory_pools(): break
    except Exception: pass
    if time.time() > deadline: raise SystemExit("no pools")
    time.sleep(0.3)
print("pools:", [(p.pool_id, p.storage_class.name) for p in c.memory_pools()], flush=True)

gcl = bb.GpuClient(c, 0); gcl.init()
N, S = 4, 1 * MB
src = g.malloc(N * S); dst = g.malloc(N * S)
g.fill_pattern(src, N * S, seed=3); g.sync()
items = [("sh%d" % i, src + i * S, S) for i in range(N)]
cfgp = bb.PlacementConfig(); cfgp.checksum = True

print("== fused path:", flush=True)
try:
    st = gcl.batch_put_device(items, cfgp)
    print("put statuses:", st, flush=True)
    stg = gcl.batch_get_device([(k, dst + i * S, S) for i, (k, _, _) in enumerate(items)], verify=True)
    print("get statuses:", stg, flush=True)
    print("verify:", g.verify_pattern(dst, N * S, seed=3), flush=True)
    c.batch_remove([k for k, _, _ in items])
except Exception as e:
    print("fused EXC:", e, flush=True)

print("== sdma path (fused off):", flush=True)
try:
    gcl2 = bb.GpuClient(c, 0); gcl2.init(); gcl2.set_fused_copy(False)
    st = gcl2.batch_put_device([("sd%d" % i, src + i * S, S) for i in range(N)], cfgp)
    print("put statuses:", st, flush=True)
except Exception as e:
    print("sdma EXC:", e, flush=True)

for p in procs: p.send_signal(signal.SIGTERM)
for p in procs:
    try: p.wait(timeout=5)
    except subprocess.TimeoutExpired: p.kill()
print("DONE", flush=True)
