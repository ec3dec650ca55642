"""In-process replication pull loop — the control-flow mirror of
ReplicatedDB::pullFromUpstream (replicated_db.cpp:314-433) over the C-ABI:
a follower repeatedly asks its upstream for updates past its own
LatestSequenceNumber and applies each via HandleReplicateResponse, retrying
from the durable seq on failure. Used by the replication-chain tests and
as the wiring template for a transport layer.
"""


def pull_once(upstream_db, follower_db, max_updates=50):
    """One pull round: returns number of updates applied."""
    since = follower_db.latest_seq()  # req.seq_no (replicated_db.cpp:317)
    updates = upstream_db.get_updates(since, max_updates)
    applied = 0
    for seq, ts, rep in updates:  # apply loop (replicated_db.cpp:369-383)
        if not follower_db.handle_replicate_response(rep, ts):
            break  # failed apply -> caller re-pulls from LatestSequenceNumber
        applied += 1
    return applied


def catch_up(upstream_db, follower_db, follower_engine, max_rounds=10000):
    """Pull until the follower's seq reaches the upstream's."""
    rounds = 0
    while rounds < max_rounds:
        n = pull_once(upstream_db, follower_db)
        follower_engine.flush()
        if n == 0 and follower_db.latest_seq() >= upstream_db.latest_seq():
            return True
        rounds += 1
    return False
