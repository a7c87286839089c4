"""Sentinel values placed in the data queues (parity: reference ``marker.py:11-18``).

End-of-feed is the raw ``None`` value; ``EndPartition`` separates RDD partitions so
inference can match outputs to inputs exactly per-partition.
"""


class Marker(object):
    pass


class EndPartition(Marker):
    pass
