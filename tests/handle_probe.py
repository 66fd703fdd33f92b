import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import multiprocessing as mp
import torch

def child(val, q, ev):
    from torchstore_amd import _hipstore as e
    import torch
    torch.cuda.set_device(0)
    t = torch.full((1 << 20,), float(val), device="cuda")
    h, off, size = e.ipc_export(t.data_ptr(), 0)
    q.put((val, bytes(h), off))
    ev.wait(60)  # keep memory alive

if __name__ == "__main__":
    from torchstore_amd import _hipstore as e
    torch.cuda.set_device(0)
    ctx = mp.get_context("spawn")
    q = ctx.Queue(); ev = ctx.Event()
    procs = [ctx.Process(target=child, args=(v, q, ev), daemon=True) for v in (111, 222)]
    for p in procs: p.start()
    a = q.get(timeout=60); b = q.get(timeout=60)
    print("handle A == handle B:", a[1] == b[1], "offs:", a[2], b[2])
    out = {}
    for val, h, off in (a, b):
        base = e.ipc_open(h, 0, 0)
        probe = torch.empty(1 << 20, device="cuda")
        e.copy_batch([(probe.data_ptr(), 0, base + off, 0, (1 << 20) * 4)])
        out[val] = probe[0].item()
        print(f"exporter value {val}: read back {probe[0].item()}")
    ev.set()
    ok = out.get(111) == 111.0 and out.get(222) == 222.0
    print("DATA", "OK" if ok else "WRONG")
