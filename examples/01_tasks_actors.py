"""Core task/actor walkthrough: remote functions, objects, actors,
placement groups."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

import ray_amd as ray

ray.init()


@ray.remote
def square(x):
    return x * x


@ray.remote
class Accumulator:
    def __init__(self):
        self.total = 0

    def add(self, v):
        self.total += v
        return self.total


# parallel tasks
print("squares:", ray.get([square.remote(i) for i in range(8)]))

# zero-copy objects
big = ray.put(np.arange(1_000_000))
print("sum via task:", ray.get(square.options(name="sum").remote(2)))

# ordered actor calls
acc = Accumulator.remote()
ray.get([acc.add.remote(i) for i in range(10)])
print("accumulated:", ray.get(acc.add.remote(0)))

ray.shutdown()
