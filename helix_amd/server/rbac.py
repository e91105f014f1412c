"""Organizations, teams and unified access grants (parity with the
reference's org/team/role tables and authorizeUserToResource,
server/authz.go — AccessGrants unify user/team/org access to resources).
"""
from __future__ import annotations

import time
from typing import List, Optional

from helix_amd.server.types import new_id

ROLES = ("owner", "admin", "member", "viewer")
# role ranking for "at least" checks
_RANK = {r: i for i, r in enumerate(reversed(ROLES))}

ACTIONS = {"read": "viewer", "use": "member", "write": "admin",
           "delete": "admin", "admin": "owner"}


class RBACService:
    def __init__(self, store):
        self.store = store

    # -- orgs --------------------------------------------------------------
    def create_org(self, owner: str, name: str) -> dict:
        oid = new_id("org")
        doc = {"id": oid, "name": name, "owner": owner,
               "created": time.time()}
        self.store.put("organizations", oid, doc, owner=owner)
        self.add_member(oid, owner, "owner")
        return doc

    def get_org(self, oid: str) -> Optional[dict]:
        return self.store.get("organizations", oid)

    def list_orgs_for(self, user_id: str) -> List[dict]:
        out = []
        for m in self.store.list("memberships", owner=user_id, limit=1000):
            org = self.get_org(m["org_id"])
            if org:
                out.append({**org, "role": m["role"]})
        return out

    # -- teams -------------------------------------------------------------
    def create_team(self, org_id: str, name: str) -> dict:
        tid = new_id("team")
        doc = {"id": tid, "org_id": org_id, "name": name, "members": []}
        self.store.put("teams", tid, doc, parent=org_id)
        return doc

    def add_team_member(self, team_id: str, user_id: str):
        t = self.store.get("teams", team_id)
        if t is None:
            raise KeyError(team_id)
        if user_id not in t["members"]:
            t["members"].append(user_id)
        self.store.put("teams", team_id, t, parent=t["org_id"])
        return t

    def list_teams(self, org_id: str) -> List[dict]:
        return self.store.list("teams", parent=org_id, limit=1000)

    # -- org membership ------------------------------------------------------
    def add_member(self, org_id: str, user_id: str, role: str = "member"):
        assert role in ROLES
        mid = f"{org_id}:{user_id}"
        doc = {"id": mid, "org_id": org_id, "user_id": user_id, "role": role}
        self.store.put("memberships", mid, doc, owner=user_id,
                       parent=org_id)
        return doc

    def member_role(self, org_id: str, user_id: str) -> Optional[str]:
        doc = self.store.get("memberships", f"{org_id}:{user_id}")
        return doc["role"] if doc else None

    # -- access grants -------------------------------------------------------
    def grant(self, resource_type: str, resource_id: str, role: str,
              user_id: str = "", team_id: str = "", org_id: str = "") -> dict:
        assert role in ROLES
        gid = new_id("grant")
        doc = {"id": gid, "resource_type": resource_type,
               "resource_id": resource_id, "role": role,
               "user_id": user_id, "team_id": team_id, "org_id": org_id}
        self.store.put("access_grants", gid, doc, parent=resource_id)
        return doc

    def revoke(self, grant_id: str) -> bool:
        return self.store.delete("access_grants", grant_id)

    def grants_for(self, resource_id: str) -> List[dict]:
        return self.store.list("access_grants", parent=resource_id,
                               limit=1000)

    def authorize(self, user_id: str, resource_type: str, resource_id: str,
                  action: str, resource_owner: str = "") -> bool:
        """Unified authorization (reference authorizeUserToResource):
        owner always allowed; otherwise any grant (direct, via team, via
        org membership) with sufficient role."""
        if resource_owner and user_id == resource_owner:
            return True
        needed = _RANK[ACTIONS.get(action, "admin")]
        for g in self.grants_for(resource_id):
            if g["resource_type"] != resource_type:
                continue
            if _RANK[g["role"]] < needed:
                continue
            if g.get("user_id") == user_id:
                return True
            if g.get("team_id"):
                t = self.store.get("teams", g["team_id"])
                if t and user_id in t.get("members", []):
                    return True
            if g.get("org_id"):
                if self.member_role(g["org_id"], user_id):
                    return True
        return False
