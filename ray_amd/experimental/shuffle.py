"""Push-based distributed shuffle prototype (reference:
python/ray/experimental/shuffle.py — map tasks partition their input
and PUSH partitions straight to reducer actors, so the exchange
overlaps with mapping instead of waiting for a full materialized map
stage).

The production shuffle lives in ray_amd.data (hash exchange inside the
streaming executor); this mirrors the reference's standalone utility.
"""
from __future__ import annotations

from typing import Any, Callable, List


def _ray():
    import ray_amd

    return ray_amd


def shuffle(input_blocks: List[Any], num_reducers: int,
            partition_fn: Callable[[Any, int], List[Any]],
            reduce_fn: Callable[[List[Any]], Any]):
    """Shuffle `input_blocks` into `num_reducers` outputs.

    partition_fn(block, num_reducers) -> list of num_reducers pieces;
    reduce_fn(list_of_pieces) -> reduced output for one partition.
    Mappers push each piece to its reducer actor as soon as it is cut.
    """
    ray = _ray()

    @ray.remote
    class _Reducer:
        def __init__(self):
            self.pieces: List[Any] = []

        def push(self, piece):
            self.pieces.append(piece)

        def finish(self, fn):
            return fn(self.pieces)

    reducers = [_Reducer.remote() for _ in range(num_reducers)]

    @ray.remote
    def _mapper(block, reducers, partition_fn):
        pieces = partition_fn(block, len(reducers))
        done = [r.push.remote(p) for r, p in zip(reducers, pieces)]
        _ray().get(done)  # pushed (actor inboxes), not materialized
        return True

    ray.get([_mapper.remote(b, reducers, partition_fn)
             for b in input_blocks])
    return ray.get([r.finish.remote(reduce_fn) for r in reducers])
