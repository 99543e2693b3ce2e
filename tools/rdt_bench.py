"""RDT zero-copy GPU tensor passing: latency and effective bandwidth
between two actors sharing one MI355X (hipIpc handle exchange —
reference: RDT/GPU object store)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import ray_amd as ray


@ray.remote(num_gpus=0.5)
class Producer:
    def __init__(self):
        from ray_amd.experimental.rdt import get_gpu_object_store

        self.store = get_gpu_object_store()

    def make(self, nbytes):
        t = torch.ones(nbytes // 4, dtype=torch.float32, device="cuda")
        return self.store.put(t)


@ray.remote(num_gpus=0.5)
class Consumer:
    def __init__(self):
        from ray_amd.experimental.rdt import get_gpu_object_store

        self.store = get_gpu_object_store()

    def fetch_and_touch(self, ref):
        t0 = time.perf_counter()
        t = self.store.get(ref, device="cuda")
        s = float(t[:16].sum())  # touch (maps the memory)
        torch.cuda.synchronize()
        return time.perf_counter() - t0, s

    def reduce_all(self, ref):
        """Full-tensor reduction — the data plane actually reads every
        byte through the IPC mapping."""
        t = self.store.get(ref, device="cuda")
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        s = float(t.sum())
        torch.cuda.synchronize()
        return time.perf_counter() - t0, s


def main():
    ray.init(num_gpus=1, num_cpus=4)
    p = Producer.remote()
    c = Consumer.remote()
    for mb in (1, 64, 1024):
        nbytes = mb << 20
        ref = ray.get(p.make.remote(nbytes))
        lat, _ = ray.get(c.fetch_and_touch.remote(ref))
        # second fetch = consumer cache hit
        lat2, _ = ray.get(c.fetch_and_touch.remote(ref))
        rt, s = ray.get(c.reduce_all.remote(ref))
        bw = nbytes / rt / 1e9
        print(f"{mb:5d} MiB: first-fetch {lat * 1e3:7.2f} ms  "
              f"cached {lat2 * 1e3:6.2f} ms  full-read {bw:7.0f} GB/s "
              f"(sum={s:.0f})")
    ray.shutdown()


if __name__ == "__main__":
    main()
