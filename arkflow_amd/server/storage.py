"""Hub durable store: SQLite behind an async lock.

Mirrors reference crates/arkflow-server/src/storage.rs (3,646 LoC): tables for
nodes, desired state, intents, outbox, attempts, events and audit, driven
synchronously behind a single writer (the reference uses a StorageActor mpsc
channel; here a single asyncio lock + thread executor gives the same
serialization).
"""
from __future__ import annotations

import asyncio
import json
import sqlite3
import time
import uuid
from typing import List, Optional

_SCHEMA = """
CREATE TABLE IF NOT EXISTS nodes (
  node_id TEXT PRIMARY KEY, token TEXT, registered_at REAL,
  lease_expires REAL, last_report TEXT, online INTEGER DEFAULT 1,
  labels TEXT DEFAULT NULL
);
CREATE TABLE IF NOT EXISTS desired (
  node_id TEXT, stream_id TEXT, state TEXT, generation INTEGER,
  PRIMARY KEY (node_id, stream_id)
);
CREATE TABLE IF NOT EXISTS intents (
  intent_id TEXT PRIMARY KEY, node_id TEXT, stream_id TEXT, op TEXT,
  state TEXT, created_at REAL, updated_at REAL, error TEXT,
  attempts_made INTEGER DEFAULT 0
);
CREATE TABLE IF NOT EXISTS outbox (
  outbox_id INTEGER PRIMARY KEY AUTOINCREMENT, intent_id TEXT, node_id TEXT,
  claimed INTEGER DEFAULT 0
);
CREATE TABLE IF NOT EXISTS attempts (
  attempt_id TEXT PRIMARY KEY, intent_id TEXT, node_id TEXT, command TEXT,
  state TEXT, created_at REAL, expires_at REAL, result TEXT
);
CREATE TABLE IF NOT EXISTS events (
  seq INTEGER PRIMARY KEY AUTOINCREMENT, ts REAL, node_id TEXT, kind TEXT,
  payload TEXT
);
CREATE TABLE IF NOT EXISTS audit (
  seq INTEGER PRIMARY KEY AUTOINCREMENT, ts REAL, actor TEXT, action TEXT,
  detail TEXT
);
CREATE TABLE IF NOT EXISTS rollouts (
  rollout_id TEXT PRIMARY KEY, state TEXT, config TEXT, nodes TEXT,
  position INTEGER, created_at REAL, updated_at REAL,
  prev_config TEXT DEFAULT NULL
);
"""

# pre-upgrade databases lack these columns; applied best-effort on open
_MIGRATIONS = [
    "ALTER TABLE intents ADD COLUMN attempts_made INTEGER DEFAULT 0",
    "ALTER TABLE rollouts ADD COLUMN prev_config TEXT DEFAULT NULL",
    "ALTER TABLE nodes ADD COLUMN labels TEXT DEFAULT NULL",
]


class HubStore:
    def __init__(self, path: str = ":memory:"):
        self._db = sqlite3.connect(path, check_same_thread=False)
        self._db.row_factory = sqlite3.Row
        self._db.executescript(_SCHEMA)
        for mig in _MIGRATIONS:
            try:
                self._db.execute(mig)
            except sqlite3.OperationalError:
                pass  # column already exists
        self._lock = asyncio.Lock()

    async def _run(self, fn):
        async with self._lock:
            loop = asyncio.get_running_loop()
            return await loop.run_in_executor(None, fn)

    # ---- nodes -------------------------------------------------------------
    async def upsert_node(self, node_id: str, token: str, lease_ttl: float,
                          labels: Optional[dict] = None):
        def go():
            self._db.execute(
                "INSERT INTO nodes(node_id, token, registered_at, "
                "lease_expires, online, labels) VALUES (?,?,?,?,1,?) "
                "ON CONFLICT(node_id) DO UPDATE SET token=excluded.token, "
                "lease_expires=excluded.lease_expires, online=1, "
                "labels=COALESCE(excluded.labels, nodes.labels)",
                (node_id, token, time.time(), time.time() + lease_ttl,
                 json.dumps(labels) if labels else None))
            self._db.commit()
        await self._run(go)

    async def heartbeat(self, node_id: str, lease_ttl: float) -> bool:
        def go():
            cur = self._db.execute(
                "UPDATE nodes SET lease_expires=?, online=1 WHERE node_id=?",
                (time.time() + lease_ttl, node_id))
            self._db.commit()
            return cur.rowcount > 0
        return await self._run(go)

    async def report(self, node_id: str, snapshot: dict):
        def go():
            self._db.execute(
                "UPDATE nodes SET last_report=? WHERE node_id=?",
                (json.dumps(snapshot), node_id))
            self._db.commit()
        await self._run(go)

    async def sweep_leases(self) -> List[str]:
        def go():
            now = time.time()
            rows = self._db.execute(
                "SELECT node_id FROM nodes WHERE online=1 AND "
                "lease_expires < ?", (now,)).fetchall()
            stale = [r["node_id"] for r in rows]
            if stale:
                self._db.executemany(
                    "UPDATE nodes SET online=0 WHERE node_id=?",
                    [(n,) for n in stale])
                self._db.commit()
            return stale
        return await self._run(go)

    async def nodes(self) -> List[dict]:
        def go():
            rows = self._db.execute("SELECT * FROM nodes").fetchall()
            out = []
            for r in rows:
                d = dict(r)
                d["last_report"] = json.loads(d["last_report"]) \
                    if d["last_report"] else None
                d["labels"] = json.loads(d["labels"]) if d.get("labels") \
                    else {}
                d.pop("token", None)
                out.append(d)
            return out
        return await self._run(go)

    async def node_token(self, node_id: str) -> Optional[str]:
        def go():
            r = self._db.execute(
                "SELECT token FROM nodes WHERE node_id=?",
                (node_id,)).fetchone()
            return r["token"] if r else None
        return await self._run(go)

    # ---- desired state ------------------------------------------------------
    async def set_desired(self, node_id: str, stream_id: str, state: str
                          ) -> int:
        """Record the operator's desired state; bumps the generation."""
        def go():
            r = self._db.execute(
                "SELECT generation FROM desired WHERE node_id=? AND "
                "stream_id=?", (node_id, stream_id)).fetchone()
            gen = (r["generation"] if r else 0) + 1
            self._db.execute(
                "INSERT INTO desired VALUES (?,?,?,?) ON CONFLICT"
                "(node_id, stream_id) DO UPDATE SET state=excluded.state, "
                "generation=excluded.generation",
                (node_id, stream_id, state, gen))
            self._db.commit()
            return gen
        return await self._run(go)

    async def desired(self, node_id: str) -> List[dict]:
        def go():
            rows = self._db.execute(
                "SELECT * FROM desired WHERE node_id=?",
                (node_id,)).fetchall()
            return [dict(r) for r in rows]
        return await self._run(go)

    # ---- intents / outbox / attempts ----------------------------------------
    async def enqueue_intent(self, node_id: str, stream_id: str, op: str
                             ) -> str:
        intent_id = uuid.uuid4().hex[:12]

        def go():
            now = time.time()
            self._db.execute(
                "INSERT INTO intents VALUES (?,?,?,?,?,?,?,NULL,0)",
                (intent_id, node_id, stream_id, op, "pending", now, now))
            self._db.execute(
                "INSERT INTO outbox(intent_id, node_id) VALUES (?,?)",
                (intent_id, node_id))
            self._db.commit()
        await self._run(go)
        return intent_id

    async def claim_outbox(self, online: List[str],
                           limit: int = 16) -> List[dict]:
        """Claim dispatchable outbox rows — ONLY for nodes currently online
        with a valid lease. Rows for offline nodes stay claimable so the
        intent executes when the node returns (reference hub.rs:468-520
        keeps outbox rows pending until the node holds a valid lease)."""
        def go():
            if not online:
                return []
            marks = ",".join("?" for _ in online)
            rows = self._db.execute(
                "SELECT o.outbox_id, o.intent_id, o.node_id, i.stream_id, "
                f"i.op FROM outbox o JOIN intents i USING (intent_id) "
                f"WHERE o.claimed=0 AND o.node_id IN ({marks}) LIMIT ?",
                (*online, limit)).fetchall()
            for r in rows:
                self._db.execute(
                    "UPDATE outbox SET claimed=1 WHERE outbox_id=?",
                    (r["outbox_id"],))
            self._db.commit()
            return [dict(r) for r in rows]
        return await self._run(go)

    async def create_attempt(self, intent_id: str, node_id: str,
                             command: dict, ttl: float = 60.0) -> str:
        attempt_id = uuid.uuid4().hex[:12]

        def go():
            now = time.time()
            self._db.execute(
                "INSERT INTO attempts VALUES (?,?,?,?,?,?,?,NULL)",
                (attempt_id, intent_id, node_id, json.dumps(command),
                 "pending", now, now + ttl))
            self._db.execute(
                "UPDATE intents SET state='dispatched', updated_at=?, "
                "attempts_made=attempts_made+1 WHERE intent_id=?",
                (now, intent_id))
            self._db.commit()
        await self._run(go)
        return attempt_id

    async def pending_commands(self, node_id: str) -> List[dict]:
        def go():
            rows = self._db.execute(
                "SELECT attempt_id, command FROM attempts WHERE node_id=? "
                "AND state='pending' AND expires_at > ?",
                (node_id, time.time())).fetchall()
            return [{"attempt_id": r["attempt_id"],
                     **json.loads(r["command"])} for r in rows]
        return await self._run(go)

    async def command_result(self, attempt_id: str, ok: bool,
                             detail: str = "") -> Optional[str]:
        def go():
            r = self._db.execute(
                "SELECT intent_id FROM attempts WHERE attempt_id=?",
                (attempt_id,)).fetchone()
            if r is None:
                return None
            now = time.time()
            self._db.execute(
                "UPDATE attempts SET state=?, result=? WHERE attempt_id=?",
                ("succeeded" if ok else "failed", detail, attempt_id))
            self._db.execute(
                "UPDATE intents SET state=?, updated_at=?, error=? "
                "WHERE intent_id=?",
                ("succeeded" if ok else "failed", now,
                 None if ok else detail, r["intent_id"]))
            self._db.commit()
            return r["intent_id"]
        return await self._run(go)

    async def expire_attempts(self, max_retries: int = 5) -> int:
        """Expire timed-out pending attempts and RE-ENQUEUE their intents
        (an outbox row becomes claimable again) until the retry budget runs
        out, then fail the intent. Regression for the r1 hole where expired
        attempts left intents stuck at 'dispatched' forever (reference
        hub.rs:411 attempt expiry feeds reconciliation)."""
        def go():
            now = time.time()
            rows = self._db.execute(
                "SELECT attempt_id, intent_id FROM attempts WHERE "
                "state='pending' AND expires_at < ?", (now,)).fetchall()
            for r in rows:
                self._db.execute(
                    "UPDATE attempts SET state='expired' WHERE attempt_id=?",
                    (r["attempt_id"],))
                it = self._db.execute(
                    "SELECT node_id, attempts_made FROM intents WHERE "
                    "intent_id=?", (r["intent_id"],)).fetchone()
                if it is None:
                    continue  # rollout attempt — handled by advance_rollouts
                if it["attempts_made"] >= max_retries:
                    self._db.execute(
                        "UPDATE intents SET state='failed', updated_at=?, "
                        "error='attempt retries exhausted' WHERE intent_id=?",
                        (now, r["intent_id"]))
                else:
                    self._db.execute(
                        "UPDATE intents SET state='pending', updated_at=? "
                        "WHERE intent_id=?", (now, r["intent_id"]))
                    self._db.execute(
                        "INSERT INTO outbox(intent_id, node_id) VALUES (?,?)",
                        (r["intent_id"], it["node_id"]))
            self._db.commit()
            return len(rows)
        return await self._run(go)

    async def attempt_for_intent(self, intent_id: str) -> Optional[dict]:
        """Latest attempt row for an intent (rollout stages use synthetic
        intent ids, so this is how rollout health gating reads results)."""
        def go():
            r = self._db.execute(
                "SELECT * FROM attempts WHERE intent_id=? "
                "ORDER BY created_at DESC LIMIT 1", (intent_id,)).fetchone()
            return dict(r) if r else None
        return await self._run(go)

    async def intents(self, limit: int = 100) -> List[dict]:
        def go():
            rows = self._db.execute(
                "SELECT * FROM intents ORDER BY created_at DESC LIMIT ?",
                (limit,)).fetchall()
            return [dict(r) for r in rows]
        return await self._run(go)

    # ---- events / audit -------------------------------------------------------
    async def push_event(self, node_id: str, kind: str, payload: dict):
        def go():
            self._db.execute(
                "INSERT INTO events(ts, node_id, kind, payload) "
                "VALUES (?,?,?,?)",
                (time.time(), node_id, kind, json.dumps(payload)))
            self._db.commit()
        await self._run(go)

    async def events(self, after_seq: int = 0, limit: int = 100) -> List[dict]:
        def go():
            rows = self._db.execute(
                "SELECT * FROM events WHERE seq > ? ORDER BY seq LIMIT ?",
                (after_seq, limit)).fetchall()
            return [{**dict(r), "payload": json.loads(r["payload"])}
                    for r in rows]
        return await self._run(go)

    async def audit(self, actor: str, action: str, detail: str = ""):
        def go():
            self._db.execute(
                "INSERT INTO audit(ts, actor, action, detail) VALUES (?,?,?,?)",
                (time.time(), actor, action, detail))
            self._db.commit()
        await self._run(go)

    async def audit_log(self, limit: int = 100) -> List[dict]:
        def go():
            rows = self._db.execute(
                "SELECT * FROM audit ORDER BY seq DESC LIMIT ?",
                (limit,)).fetchall()
            return [dict(r) for r in rows]
        return await self._run(go)

    # ---- rollouts --------------------------------------------------------------
    async def create_rollout(self, config: dict, nodes: List[str],
                             prev_config: Optional[dict] = None) -> str:
        rid = uuid.uuid4().hex[:12]

        def go():
            now = time.time()
            self._db.execute(
                "INSERT INTO rollouts VALUES (?,?,?,?,?,?,?,?)",
                (rid, "running", json.dumps(config), json.dumps(nodes), 0,
                 now, now,
                 json.dumps(prev_config) if prev_config is not None
                 else None))
            self._db.commit()
        await self._run(go)
        return rid

    async def get_rollout(self, rid: str) -> Optional[dict]:
        def go():
            r = self._db.execute(
                "SELECT * FROM rollouts WHERE rollout_id=?", (rid,)).fetchone()
            if r is None:
                return None
            d = dict(r)
            d["config"] = json.loads(d["config"])
            d["nodes"] = json.loads(d["nodes"])
            d["prev_config"] = json.loads(d["prev_config"]) \
                if d.get("prev_config") else None
            return d
        return await self._run(go)

    async def update_rollout(self, rid: str, state: Optional[str] = None,
                             position: Optional[int] = None):
        def go():
            if state is not None:
                self._db.execute(
                    "UPDATE rollouts SET state=?, updated_at=? "
                    "WHERE rollout_id=?", (state, time.time(), rid))
            if position is not None:
                self._db.execute(
                    "UPDATE rollouts SET position=?, updated_at=? "
                    "WHERE rollout_id=?", (position, time.time(), rid))
            self._db.commit()
        await self._run(go)

    async def rollouts(self) -> List[dict]:
        def go():
            rows = self._db.execute("SELECT * FROM rollouts").fetchall()
            return [dict(r) for r in rows]
        return await self._run(go)

    def close(self):
        self._db.close()
