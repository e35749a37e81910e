"""ws=1 RCCL graph-capture canary: is ncclAllReduce capturable on this
stack (torch 2.10 + RCCL 2.26 + ROCm 7.2)?  Informs the multi-GPU
graph-capture path (solvers/hip.py body_dist); its fallback makes a
'no' harmless."""
import os, torch, torch.distributed as dist

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29533")
dist.init_process_group("nccl", rank=0, world_size=1)
torch.cuda.set_device(0)
t = torch.ones(4, dtype=torch.float64, device="cuda")
buf = torch.zeros(8, dtype=torch.float64, device="cuda")
dist.all_reduce(t)  # warm comm
s1 = torch.cuda.Stream()
ev = torch.cuda.Event()
try:
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        cur = torch.cuda.current_stream()
        ev.record(cur)
        s1.wait_event(ev)
        ev2 = torch.cuda.Event()
        with torch.cuda.stream(s1):
            dist.all_reduce(t)
            ev2.record(s1)
        cur.wait_event(ev2)
        buf += t.sum()
    for _ in range(3):
        g.replay()
    torch.cuda.synchronize()
    print("CAPTURE-OK", float(t[0]), float(buf[0]))
except Exception as e:
    print("CAPTURE-FAIL", type(e).__name__, str(e)[:200])

# P2P (halo shape): grouped isend/irecv to self under capture.
# KNOWN: at world_size=1 NCCL hard-aborts the process on a SELF-send
# (invalid usage) before the capture question arises -- this section only
# yields data on a >=2-GPU node.  The solver's capture fallback makes a
# P2P-capture 'no' harmless either way.
try:
    src = torch.arange(4, dtype=torch.float64, device="cuda")
    dst = torch.zeros(4, dtype=torch.float64, device="cuda")
    ops = [dist.P2POp(dist.irecv, dst, 0), dist.P2POp(dist.isend, src, 0)]
    for r in dist.batch_isend_irecv(ops):
        r.wait()
    torch.cuda.synchronize()  # warm
    g2 = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g2):
        ops = [dist.P2POp(dist.irecv, dst, 0), dist.P2POp(dist.isend, src, 0)]
        for r in dist.batch_isend_irecv(ops):
            r.wait()
    src += 1.0
    g2.replay()
    torch.cuda.synchronize()
    print("P2P-CAPTURE-OK", dst.tolist())
except Exception as e:
    print("P2P-CAPTURE-FAIL", type(e).__name__, str(e)[:200])
dist.destroy_process_group()
