"""Head-node HA: leader election for standby heads.

Role parity: ant fork python/ray/ha/ (redis_leader_selector.py —
RedisBasedLeaderSelector: standby head processes campaign on a Redis lease
key; the winner starts GCS, losers poll). Offline-image build: the same
protocol over a lockfile lease (FileLeaderSelector); RedisBasedLeaderSelector
keeps the reference API and activates when a redis client+server exist.
"""
from ant_ray_amd.ha.leader_selector import (  # noqa: F401
    FileLeaderSelector,
    LeaderSelector,
    RedisBasedLeaderSelector,
)
