"""Probe: native RcclComm with 2 ranks on ONE GPU (RCCL may refuse
duplicate devices like NCCL does — this records the answer)."""
import multiprocessing as mp
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def worker(rank, uid, q):
    try:
        import torch

        torch.cuda.set_device(0)
        from ray_amd import _rccl_comm as rc

        comm = rc.RcclComm(2, rank, uid, 0)
        t = torch.ones(64, device="cuda") * (rank + 1)
        comm.allreduce(t, "sum")
        comm.synchronize()
        q.put((rank, "ok", float(t[0].item())))
    except Exception as e:
        q.put((rank, "fail", str(e)[:300]))


def main():
    from ray_amd import _rccl_comm as rc

    print("rccl version:", rc.version())
    uid = rc.unique_id()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=worker, args=(r, uid, q)) for r in range(2)]
    for p in ps:
        p.start()
    results = []
    for _ in range(2):
        try:
            results.append(q.get(timeout=120))
        except Exception:
            results.append(("?", "timeout", ""))
            break
    for p in ps:
        p.join(timeout=10)
        if p.is_alive():
            p.terminate()
    print("2rank-1gpu:", results)


if __name__ == "__main__":
    main()
