"""In-process replication pull loop — the control-flow mirror of
ReplicatedDB::pullFromUpstream (replicated_db.cpp:314-433) over the C-ABI:
a follower repeatedly asks its upstream for updates past its own
LatestSequenceNumber and applies each via HandleReplicateResponse, retrying
from the durable seq on failure. Used by the replication-chain tests and
as the wiring template for a transport layer.
"""
import threading


def pull_once(upstream_db, follower_db, max_updates=50, observer=False):
    """One pull round: returns number of updates applied."""
    since = follower_db.latest_seq()  # req.seq_no (replicated_db.cpp:317)
    try:
        updates = upstream_db.get_updates(since, max_updates,
                                          observer=observer)
    except TypeError:  # transports without the role parameter
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


LEADER = "LEADER"
FOLLOWER = "FOLLOWER"
OBSERVER = "OBSERVER"


class WriteToSlaveError(RuntimeError):
    """≅ ReturnCode::WRITE_TO_SLAVE (rocksdb_replicator.h:71-78; thrown by
    ReplicatedDB::Write on a follower, replicated_db.cpp:107-109)."""


class ReplicatedShard:
    """One registered db ≅ a ReplicatedDB: role, upstream, pull thread."""

    def __init__(self, name, db, role, upstream_db=None):
        self.name = name
        self.db = db
        self.role = role
        self.upstream_db = upstream_db
        self._stop = threading.Event()
        self._thread = None
        # write-degradation state is PER ReplicatedDB in the reference
        # (replicated_db.cpp:236-273 keeps it on the instance): one shard's
        # healthy acks must not reset another's degradation, and concurrent
        # write() calls on this shard serialize the counter under a lock
        self._miss_mu = threading.Lock()
        self._consecutive_misses = 0


class Replicator:
    """≅ RocksDBReplicator (rocksdb_replicator.h:160-256) re-imagined for the
    GPU engine: a per-process registry of named replicated shards over one
    GraEngine per GPU, with per-shard roles, an in-process pull loop per
    FOLLOWER/OBSERVER (the pullFromUpstream control flow), write modes 0/1/2
    with follower-ACK waits (MaxNumberBox equivalent in the engine), and the
    reference's write-degradation behavior (replicated_db.cpp:236-273).

    The registry itself is the FastReadMap's role (read-mostly name->db
    lookup, fast_read_map.h:36-140); in this control plane a dict under a
    lock suffices — the data plane never touches it.
    """

    ACK_TIMEOUT_MS = 2000        # replicated_db.cpp:66-67
    DEGRADED_TIMEOUT_MS = 10     # :70-73
    DEGRADE_AFTER_MISSES = 100   # :75-78

    def __init__(self, engine, pull_interval_s=0.002):
        self.engine = engine
        self.pull_interval_s = pull_interval_s
        self._mu = threading.Lock()
        self._dbs = {}
        self._next_shard = 0

    # ≅ RocksDBReplicator::addDB (rocksdb_replicator.cpp:96-133)
    def add_db(self, name, role, upstream_db=None):
        with self._mu:
            if name in self._dbs:
                raise KeyError(f"db exists: {name}")
            shard = self._next_shard
            self._next_shard += 1
            db = self.engine.open(shard)
            rs = ReplicatedShard(name, db, role, upstream_db)
            self._dbs[name] = rs
        if role in (FOLLOWER, OBSERVER) and upstream_db is not None:
            self._start_pull(rs)  # FOLLOWER starts pulling immediately (:126-128)
        return rs

    # ≅ RocksDBReplicator::removeDB (rocksdb_replicator.cpp:135-154)
    def remove_db(self, name):
        with self._mu:
            rs = self._dbs.pop(name)
        self._stop_pull(rs)
        rs.db.close()

    def get(self, name):
        with self._mu:
            return self._dbs[name]

    # ≅ changeDBRoleAndUpstream (admin_handler.cpp:1446-1484 -> addDB role
    # transition): promote/demote in place
    def change_role(self, name, role, upstream_db=None):
        rs = self.get(name)
        self._stop_pull(rs)
        rs.role = role
        rs.upstream_db = upstream_db
        if role in (FOLLOWER, OBSERVER) and upstream_db is not None:
            self._start_pull(rs)

    # ≅ RocksDBReplicator::write -> ReplicatedDB::Write
    # (replicated_db.cpp:103-166)
    def write(self, name, rep_bytes, mode=0):
        rs = self.get(name)
        if rs.role != LEADER:
            raise WriteToSlaveError(name)  # :107-109
        seq = rs.db.write_leader(rep_bytes)
        if mode in (1, 2):
            with rs._miss_mu:
                degraded = rs._consecutive_misses >= self.DEGRADE_AFTER_MISSES
            timeout = (self.DEGRADED_TIMEOUT_MS if degraded
                       else self.ACK_TIMEOUT_MS)
            ok = rs.db.wait_ack(seq, confirmed=(mode == 2), timeout_ms=timeout)
            with rs._miss_mu:
                if ok:
                    rs._consecutive_misses = 0
                else:
                    rs._consecutive_misses += 1  # degradation (:236-273)
        return seq

    def _start_pull(self, rs):
        rs._stop.clear()

        def loop():
            while not rs._stop.is_set():
                try:
                    n = pull_once(rs.upstream_db, rs.db,
                                  observer=(rs.role == OBSERVER))
                    if n:
                        self.engine.flush()
                        continue  # more may be pending: immediate re-pull (:430)
                except RuntimeError:
                    pass  # retry after delay (randomized in the reference, :412-431)
                rs._stop.wait(self.pull_interval_s)

        rs._thread = threading.Thread(target=loop, daemon=True,
                                      name=f"rptor-{rs.name}")
        rs._thread.start()

    def _stop_pull(self, rs):
        if rs._thread:
            rs._stop.set()
            rs._thread.join(timeout=10)
            rs._thread = None

    def close(self):
        """Stops this replicator's pull threads and closes its handles.
        LIFETIME RULE: close downstream replicators BEFORE their upstreams —
        a pull thread holds its upstream's db handle (the reference keeps
        ReplicatedDB alive by shared_ptr until pullers drop,
        rocksdb_replicator.cpp:145-151; C handles have no refcount)."""
        with self._mu:
            dbs = list(self._dbs.values())
            self._dbs.clear()
        for rs in dbs:
            self._stop_pull(rs)
            rs.db.close()


def dump_stats_text(replicator):
    """Text stats dump ≅ common::Stats::DumpStatsAsText with the reference's
    per-db counter naming scheme (replicator_stats.cpp:33-102 +
    per-db suffixing as in replicator_stats.h:66-72)."""
    lines = []
    with replicator._mu:
        items = list(replicator._dbs.items())
    for name, rs in items:
        c = rs.db.counters()
        avg_lat = (c["latency_ms_sum"] / c["latency_samples"]
                   if c["latency_samples"] else 0.0)
        lines += [
            f"replicator_in_updates_{name}: {c['updates_applied']}",
            f"replicator_in_bytes_{name}: {c['in_bytes']}",
            f"replicator_out_updates_{name}: {c['updates_served']}",
            f"replicator_out_bytes_{name}: {c['out_bytes']}",
            f"replicator_handle_response_failure_{name}: {c['apply_failures']}",
            f"replicator_latency_ms_{name}: {avg_lat:.3f}",
            f"latest_seq_{name}: {c['latest_seq']}",
            f"role_{name}: {rs.role}",
        ]
    return "\n".join(lines)
