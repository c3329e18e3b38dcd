"""Data-parallel stream sharding (DP over topic partitions).

The reference partitions streams by Kafka topic partition and keys Flink
state by PARTITION BY columns.  Here each rank owns a
static subset of partitions; keyed operator state (windows, joins, anomaly
history) lives with the owning rank, so the steady-state pipeline needs
ZERO cross-rank traffic — collectives appear only in retrieval
(shard_index), TP, and metric reduction.

Key->partition uses murmur2 like the Java Kafka client so records land on
the same partition a real producer would choose.
"""

from __future__ import annotations


def murmur2(data: bytes) -> int:
    """Kafka's murmur2 (positive 31-bit), for default key partitioning."""
    length = len(data)
    seed = 0x9747B28C
    m = 0x5BD1E995
    r = 24
    h = (seed ^ length) & 0xFFFFFFFF
    i = 0
    while length >= 4:
        k = (data[i] | (data[i + 1] << 8) | (data[i + 2] << 16)
             | (data[i + 3] << 24))
        k = (k * m) & 0xFFFFFFFF
        k ^= k >> r
        k = (k * m) & 0xFFFFFFFF
        h = (h * m) & 0xFFFFFFFF
        h ^= k
        i += 4
        length -= 4
    if length >= 3:
        h ^= data[i + 2] << 16
    if length >= 2:
        h ^= data[i + 1] << 8
    if length >= 1:
        h ^= data[i]
        h = (h * m) & 0xFFFFFFFF
    # finalization runs unconditionally (Java Utils.murmur2 applies
    # h ^= h >>> 13 after the tail switch, even for length % 4 == 0)
    h ^= h >> 13
    h = (h * m) & 0xFFFFFFFF
    h ^= h >> 15
    return h & 0x7FFFFFFF


def partition_for_key(key: str | bytes, n_partitions: int) -> int:
    if isinstance(key, str):
        key = key.encode()
    return murmur2(key) % n_partitions


class PartitionAssignment:
    """Static round-robin assignment of topic partitions to DP ranks."""

    def __init__(self, n_partitions: int, world_size: int = 1, rank: int = 0):
        assert n_partitions >= world_size, \
            "need at least one partition per rank"
        self.n_partitions = n_partitions
        self.world_size = world_size
        self.rank = rank

    def owner(self, partition: int) -> int:
        return partition % self.world_size

    def mine(self, partition: int) -> bool:
        return self.owner(partition) == self.rank

    @property
    def owned(self) -> list[int]:
        return [p for p in range(self.n_partitions) if self.mine(p)]

    def owns_key(self, key: str | bytes) -> bool:
        return self.mine(partition_for_key(key, self.n_partitions))

    def filter_records(self, records: list[dict],
                       key_field: str) -> list[dict]:
        return [r for r in records if self.owns_key(str(r[key_field]))]
