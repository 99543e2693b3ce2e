import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import ray_amd as ray

for trial in range(12):
    ray.init(num_cpus=4, num_gpus=1, ignore_reinit_error=False)
    try:
        @ray.remote(num_gpus=0.3, tensor_transport="hipipc")
        class Prod:
            def __init__(self):
                self.t = None
            def make(self, n):
                self.t = torch.ones(n, device="cuda")
                torch.cuda.synchronize()
                return self.t
            def own_sum(self):
                torch.cuda.synchronize()
                return float(self.t.sum())

        @ray.remote(num_gpus=0.3)
        class Cons:
            def __init__(self):
                self.got = None
            def recv(self, t):
                self.got = t
                s1 = float(t.sum())
                torch.cuda.synchronize()
                s2 = float(t.sum())
                # fresh re-map via a new fetch would need the ref; just re-read
                time.sleep(0.05)
                s3 = float(self.got.sum())
                return s1, s2, s3, str(t.dtype), t.shape[0], float(t[:4].sum())

        p = Prod.remote()
        c = Cons.remote()
        n = 1024
        ref = p.make.remote(n)
        s1, s2, s3, dt, ln, head = ray.get(c.recv.remote(ref), timeout=120)
        psum = ray.get(p.own_sum.remote(), timeout=60)
        ok = s1 == float(n)
        print(f"trial {trial}: consumer {s1} / resync {s2} / later {s3} "
              f"head4={head} producer={psum} {'OK' if ok else '<<< BAD'}")
    finally:
        ray.shutdown()
