"""Step-by-step probe: RCCL ws=1 collectives, then inside hipGraph capture.

Prints a line (flushed) after every step so the hang point is visible in
the log even when the process is killed by timeout.
"""
import os
import sys
import torch
import torch.distributed as dist

print("step0 torch", torch.__version__, flush=True)
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29661")
dist.init_process_group("nccl", init_method="tcp://127.0.0.1:29662",
                        world_size=1, rank=0)
print("step1 init_process_group ok", flush=True)
dev = torch.device("cuda:0")
torch.cuda.set_device(dev)
send = torch.ones(1 << 20, device=dev)
gath = torch.empty(1 << 20, device=dev)
dist.all_gather_into_tensor(gath, send)
torch.cuda.synchronize()
print("step2 eager all_gather ok", gath.mean().item(), flush=True)
a2a_s = torch.arange(1024, dtype=torch.float32, device=dev)
a2a_r = torch.empty_like(a2a_s)
dist.all_to_all_single(a2a_r, a2a_s)
red = torch.full((4096,), 2.0, device=dev)
dist.all_reduce(red)
torch.cuda.synchronize()
print("step3 eager a2a+allreduce ok", flush=True)

send.fill_(3.0)
g = torch.cuda.CUDAGraph()
print("step4 begin capture", flush=True)
try:
    with torch.cuda.graph(g):
        dist.all_gather_into_tensor(gath, send)
    print("step5 capture(all_gather) ok", flush=True)
except Exception as e:
    print("step5 capture FAILED:", repr(e), flush=True)
    sys.exit(1)
gath.zero_()
torch.cuda.synchronize()
print("step6 pre-replay sync ok", flush=True)
g.replay()
torch.cuda.synchronize()
print("step7 replay ok mean=", gath.mean().item(), flush=True)

g2 = torch.cuda.CUDAGraph()
with torch.cuda.graph(g2):
    dist.all_to_all_single(a2a_r, a2a_s)
    dist.all_reduce(red)
g2.replay()
torch.cuda.synchronize()
print("step8 capture+replay a2a/allreduce ok", flush=True)
print("step9 ALL OK", flush=True)
# dist.destroy_process_group() deadlocks under a pytest parent (observed:
# step8 prints, destroy never returns — gpurun_out/r2_call13.log; clean
# standalone).  The probe's work is done and verified; exit hard and let
# the OS reclaim the communicator.
os._exit(0)
