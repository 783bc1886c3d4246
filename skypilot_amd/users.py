"""Users, roles and service accounts (RBAC).

Reference: sky/users/rbac.py + sky/users/server.py (casbin-backed
admin/user roles, service-account tokens with `sky api` auth), collapsed
to the pool model: roles live in the users table, service accounts are
hashed bearer tokens, and authorization is a per-request-type check
applied by the API server before a request is scheduled.

Roles:
  admin  — everything, including user/role/token management and
           operating other users' clusters.
  user   — full use of the pool, but only their own clusters for
           stop/down/start/autostop/cancel.
  viewer — read-only request types.
"""
from __future__ import annotations

import contextlib
import hashlib
import secrets
import sqlite3
import time
from typing import Any, Dict, List, Optional

from skypilot_amd import global_state
from skypilot_amd.exceptions import PermissionDeniedError

ROLES = ("admin", "user", "viewer")

# Read-only request types (viewer-allowed).  Everything else requires
# role >= user; user management is handled by admin-only API routes.
READONLY_REQUESTS = {
    "status", "queue", "job_status", "check", "show_gpus",
    "cluster_events", "storage_list", "volumes_list", "recipes_list",
    "jobs_queue", "jobs_logs", "jobs_pool_status", "jobs_group_status",
    "serve_status", "serve_logs", "cost_report",
}

_SCHEMA = """
CREATE TABLE IF NOT EXISTS service_accounts (
    token_hash TEXT PRIMARY KEY,
    name TEXT NOT NULL,
    owner TEXT NOT NULL,
    role TEXT NOT NULL,
    created_at REAL,
    last_used REAL
);
"""


@contextlib.contextmanager
def _conn():
    conn = sqlite3.connect(global_state.root_dir() / "users.db",
                           timeout=30)
    try:
        conn.execute("PRAGMA busy_timeout=30000")
        conn.execute("PRAGMA journal_mode=WAL")
        conn.executescript(_SCHEMA)
        with conn:
            yield conn
    finally:
        conn.close()


def _users_conn():
    # roles live in global_state's users table (added column)
    return global_state._conn()  # noqa: SLF001 (same package)


def _ensure_role_column(c) -> None:
    cols = [r[1] for r in c.execute("PRAGMA table_info(users)")]
    if "role" not in cols:
        c.execute("ALTER TABLE users ADD COLUMN role TEXT DEFAULT 'user'")


def ensure_user(name: str, role: Optional[str] = None) -> str:
    """Create the user if missing; returns its role.  The first user
    ever seen (the API server's own identity) bootstraps as admin."""
    with _users_conn() as c:
        _ensure_role_column(c)
        row = c.execute("SELECT role FROM users WHERE name=?",
                        (name,)).fetchone()
        if row is not None:
            return row[0] or "user"
        if role is None:
            n_admin = c.execute(
                "SELECT COUNT(*) FROM users WHERE role='admin'"
            ).fetchone()[0]
            role = "admin" if n_admin == 0 else "user"
        c.execute("INSERT INTO users (name, created_at, role) "
                  "VALUES (?,?,?)", (name, time.time(), role))
        return role


def get_role(name: str) -> str:
    return ensure_user(name)


def set_role(name: str, role: str) -> None:
    if role not in ROLES:
        raise ValueError(f"role must be one of {ROLES}")
    ensure_user(name)
    with _users_conn() as c:
        _ensure_role_column(c)
        c.execute("UPDATE users SET role=? WHERE name=?", (role, name))


def list_users() -> List[Dict[str, Any]]:
    with _users_conn() as c:
        _ensure_role_column(c)
        rows = c.execute(
            "SELECT name, role, created_at FROM users").fetchall()
    return [{"name": r[0], "role": r[1] or "user", "created_at": r[2]}
            for r in rows]


# ---------------------------------------------------------------- tokens
def _hash(token: str) -> str:
    return hashlib.sha256(token.encode()).hexdigest()


def create_token(name: str, owner: str, role: str = "user") -> str:
    """Mint a service-account token.  The plaintext is returned exactly
    once; only its sha256 is stored."""
    if role not in ROLES:
        raise ValueError(f"role must be one of {ROLES}")
    token = "sky_" + secrets.token_hex(24)
    with _conn() as c:
        c.execute(
            "INSERT INTO service_accounts "
            "(token_hash,name,owner,role,created_at) VALUES (?,?,?,?,?)",
            (_hash(token), name, owner, role, time.time()))
    return token


def resolve_token(token: str) -> Optional[Dict[str, Any]]:
    with _conn() as c:
        row = c.execute(
            "SELECT name, role FROM service_accounts WHERE token_hash=?",
            (_hash(token),)).fetchone()
        if row is None:
            return None
        c.execute("UPDATE service_accounts SET last_used=? "
                  "WHERE token_hash=?", (time.time(), _hash(token)))
    return {"name": f"sa:{row[0]}", "role": row[1]}


def list_tokens() -> List[Dict[str, Any]]:
    with _conn() as c:
        rows = c.execute(
            "SELECT name, owner, role, created_at, last_used "
            "FROM service_accounts").fetchall()
    return [{"name": r[0], "owner": r[1], "role": r[2],
             "created_at": r[3], "last_used": r[4]} for r in rows]


def revoke_token(name: str) -> int:
    with _conn() as c:
        cur = c.execute("DELETE FROM service_accounts WHERE name=?",
                        (name,))
        return cur.rowcount


# ---------------------------------------------------------- authorization
def authorize(role: str, request_name: str) -> None:
    """Raise PermissionDeniedError if `role` may not schedule
    `request_name`."""
    if role == "admin":
        return
    if role == "viewer" and request_name not in READONLY_REQUESTS:
        raise PermissionDeniedError(
            f"role 'viewer' may not run {request_name!r}")
    # role 'user': everything schedulable; per-cluster ownership is
    # enforced by check_cluster_owner at execution time.


def authorize_workspace(user: str, role: str, workspace: str) -> None:
    """Workspace-membership RBAC (reference: sky workspaces with
    `private: true` + allowed_users).  Config:

        workspaces:
          team-a:
            private: true
            allowed_users: [alice, bob]

    Unlisted workspaces (or non-private ones) are open; admins always
    pass."""
    if role == "admin" or not workspace:
        return
    from skypilot_amd import config as sky_config
    spec = sky_config.get_nested(["workspaces", workspace], None)
    if not isinstance(spec, dict) or not spec.get("private"):
        return
    allowed = spec.get("allowed_users") or []
    if user not in allowed:
        raise PermissionDeniedError(
            f"workspace {workspace!r} is private; user {user!r} is not "
            f"in allowed_users")


def check_cluster_owner(cluster_name: str) -> None:
    """Mutating per-cluster ops: owner or admin only."""
    rec = global_state.get_cluster(cluster_name)
    if rec is None:
        return  # let the op raise its own not-found error
    me = global_state.current_user()
    owner = rec.get("user") or me
    if owner != me and get_role(me) != "admin":
        raise PermissionDeniedError(
            f"cluster {cluster_name!r} is owned by {owner!r}; "
            f"{me!r} (role {get_role(me)!r}) may not modify it")
