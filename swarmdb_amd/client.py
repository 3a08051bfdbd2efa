"""SwarmDBClient — Python client for the REST API.

The reference documents raw curl usage only (reference README.md:102+);
this client wraps the same endpoints with typed helpers. Works against
any server exposing the compatible surface (this framework or the
reference's intended API).

Usage::

    from swarmdb_amd.client import SwarmDBClient

    with SwarmDBClient("http://localhost:8000", agent_id="agent1") as c:
        c.register()
        mid = c.send("agent2", "hello")
        for m in c.receive(timeout=1.0):
            print(m["content"])
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Union

import httpx


class SwarmDBClient:
    def __init__(
        self,
        base_url: str,
        agent_id: str,
        password: str = "password",
        timeout: float = 30.0,
    ):
        self.agent_id = agent_id
        self._http = httpx.Client(base_url=base_url, timeout=timeout)
        self._password = password
        self._token: Optional[str] = None

    # ---- lifecycle ----

    def close(self) -> None:
        self._http.close()

    def __enter__(self) -> "SwarmDBClient":
        return self

    def __exit__(self, *exc) -> None:
        self.close()

    # ---- auth ----

    def login(self) -> str:
        r = self._http.post(
            "/auth/token",
            json={"username": self.agent_id, "password": self._password},
        )
        r.raise_for_status()
        self._token = r.json()["access_token"]
        return self._token

    @property
    def _headers(self) -> Dict[str, str]:
        if self._token is None:
            self.login()
        return {"Authorization": f"Bearer {self._token}"}

    def _req(self, method: str, path: str, **kw) -> Any:
        r = self._http.request(method, path, headers=self._headers, **kw)
        if r.status_code == 401:  # token expired: refresh once
            self.login()
            r = self._http.request(method, path, headers=self._headers, **kw)
        if r.status_code == 307:
            # distributed gateway: the agent lives on another rank —
            # follow to its server, re-sending auth (httpx would strip
            # Authorization on a cross-host redirect; the JWT is valid
            # on every rank)
            r = self._http.request(
                method, r.headers["location"], headers=self._headers, **kw
            )
        r.raise_for_status()
        return r.json()

    # ---- agents ----

    def register(self, description: Optional[str] = None,
                 capabilities: Optional[List[str]] = None) -> dict:
        body: Dict[str, Any] = {"agent_id": self.agent_id}
        if description:
            body["description"] = description
        if capabilities:
            body["capabilities"] = capabilities
        return self._req("POST", "/agents/register", json=body)

    def deregister(self) -> dict:
        return self._req("DELETE", f"/agents/{self.agent_id}")

    # ---- messaging ----

    def send(
        self,
        receiver_id: Optional[str],
        content: Union[str, dict, list],
        message_type: str = "chat",
        priority: int = 1,
        metadata: Optional[dict] = None,
        visible_to: Optional[List[str]] = None,
    ) -> str:
        body: Dict[str, Any] = {
            "receiver_id": receiver_id,
            "content": content,
            "message_type": message_type,
            "priority": priority,
        }
        if metadata:
            body["metadata"] = metadata
        if visible_to:
            body["visible_to"] = visible_to
        return self._req("POST", "/messages", json=body)["id"]

    def send_batch(self, messages: List[dict]) -> List[str]:
        """Bulk ingestion (one engine batch server-side). Each entry:
        {receiver_id, content, message_type?, priority?}."""
        return self._req("POST", "/messages/batch", json=messages)[
            "message_ids"
        ]

    def broadcast(self, content: Union[str, dict, list],
                  exclude: Optional[List[str]] = None, **kw) -> str:
        body: Dict[str, Any] = {"content": content, **kw}
        if exclude:
            body["exclude_agents"] = exclude
        return self._req("POST", "/messages/broadcast", json=body)[
            "message_id"
        ]

    def receive(self, max_messages: int = 100, timeout: float = 1.0,
                priority_order: bool = False) -> List[dict]:
        return self._req(
            "POST",
            f"/agents/receive?max_messages={max_messages}"
            f"&timeout={timeout}&priority_order={str(priority_order).lower()}",
        )

    def get_message(self, message_id: str) -> dict:
        return self._req("GET", f"/messages/{message_id}")

    def my_messages(self, status: Optional[str] = None, limit: int = 100,
                    skip: int = 0) -> List[dict]:
        q = f"?limit={limit}&skip={skip}"
        if status:
            q += f"&status={status}"
        return self._req("GET", f"/agents/{self.agent_id}/messages{q}")

    def query(self, **params) -> List[dict]:
        q = "&".join(f"{k}={v}" for k, v in params.items() if v is not None)
        return self._req("GET", f"/messages?{q}" if q else "/messages")

    def search(self, keyword: str, case_sensitive: bool = False,
               limit: int = 100) -> List[dict]:
        return self._req(
            "GET",
            f"/messages/search/?keyword={keyword}"
            f"&case_sensitive={str(case_sensitive).lower()}&limit={limit}",
        )

    def mark_processed(self, message_id: str) -> dict:
        return self._req("PUT", f"/messages/{message_id}/status?status=processed")

    def unread_count(self) -> int:
        return self._req("GET", f"/agents/{self.agent_id}/unread_count")[
            "unread_count"
        ]

    def load(self) -> dict:
        return self._req("GET", f"/agents/{self.agent_id}/load")

    # ---- groups ----

    def create_group(self, group_name: str, agent_ids: List[str]) -> dict:
        return self._req("POST", "/groups",
                         json={"group_name": group_name,
                               "agent_ids": agent_ids})

    def send_to_group(self, group_name: str,
                      content: Union[str, dict, list], **kw) -> List[str]:
        return self._req("POST", "/groups/message",
                         json={"group_name": group_name, "content": content,
                               **kw})["message_ids"]

    # ---- LLM dispatch ----

    def dispatch_llm(self) -> str:
        return self._req("POST", "/llm/dispatch")["backend_id"]

    def complete_llm(self, backend_id: str) -> dict:
        return self._req("POST", f"/llm/complete/{backend_id}")

    # ---- admin persistence (admin token required) ----

    def admin_save(self) -> dict:
        """Reference-schema JSON snapshot."""
        return self._req("POST", "/admin/save")

    def admin_checkpoint(self) -> dict:
        """Binary full checkpoint (device-gather speed)."""
        return self._req("POST", "/admin/checkpoint")

    def admin_checkpoint_delta(self) -> dict:
        return self._req("POST", "/admin/checkpoint/delta")

    def admin_checkpoint_load(self, path: str) -> dict:
        return self._req("POST", "/admin/checkpoint/load",
                         params={"path": path})

    # ---- probes ----

    def health(self) -> dict:
        r = self._http.get("/health")
        r.raise_for_status()
        return r.json()
