"""Pluggable event queue for the worker.

The reference consumes Cloud Pub/Sub with FlowControl(max_messages=1) and
unconditional ack (worker.py:217-237) plus idempotent topic/subscription
creation helpers (pubsub_util.py). This module defines the minimal queue
interface the worker needs and two implementations:

* LocalQueue — in-process, thread-safe; used by tests and single-box
  deployments (also file-backed so a producer CLI can feed a worker
  process).
* The Pub/Sub adapter slot — constructed from env exactly like the
  reference (PROJECT / ISSUE_EVENT_TOPIC / ISSUE_EVENT_SUBSCRIPTION,
  worker.py:68-86) when google-cloud-pubsub is importable; raises a clear
  error otherwise (the library is not in this offline image)."""
from __future__ import annotations

import json
import queue
import threading
import uuid
from dataclasses import dataclass, field
from pathlib import Path
from typing import Callable, Dict, Optional


@dataclass
class Message:
    data: bytes = b""
    attributes: Dict[str, str] = field(default_factory=dict)
    message_id: str = field(default_factory=lambda: uuid.uuid4().hex)
    _acked: bool = False
    _nacked: bool = False
    _on_ack: Optional[Callable[[str], None]] = None

    def ack(self):
        self._acked = True
        if self._on_ack is not None:
            self._on_ack(self.message_id)

    def nack(self):
        self._nacked = True


class BaseQueue:
    def publish(self, data: bytes = b"", **attributes) -> str:
        raise NotImplementedError

    def pull(self, timeout: Optional[float] = None) -> Optional[Message]:
        raise NotImplementedError

    def subscribe(self, callback: Callable[[Message], None],
                  max_messages: int = 1, stop_event: Optional[threading.Event] = None,
                  poll_s: float = 0.1) -> None:
        """Pull loop, one message at a time (FlowControl(max_messages=1)
        semantics — worker.py:233-237)."""
        stop_event = stop_event or threading.Event()
        while not stop_event.is_set():
            msg = self.pull(timeout=poll_s)
            if msg is None:
                continue
            callback(msg)


class LocalQueue(BaseQueue):
    """In-memory queue; optionally mirrored to a JSONL spool file so a
    separate producer process can enqueue."""

    def __init__(self, spool_path: Optional[str] = None):
        self.q: "queue.Queue[Message]" = queue.Queue()
        self.spool = Path(spool_path) if spool_path else None
        self._spool_pos = 0
        self._ack_lock = threading.Lock()
        self._acked_ids: set = set()
        if self.spool is not None:
            # durable acks: a restarted consumer re-delivers only UNacked
            # messages (Pub/Sub at-least-once semantics, not replay-all)
            self._ack_path = self.spool.with_suffix(self.spool.suffix + ".acks")
            if self._ack_path.exists():
                self._acked_ids = set(
                    self._ack_path.read_text().split())

    def publish(self, data: bytes = b"", **attributes) -> str:
        msg = Message(data=data, attributes={k: str(v) for k, v in attributes.items()})
        if self.spool is not None:
            with open(self.spool, "a") as f:
                f.write(json.dumps({"data": data.decode("utf-8", "ignore"),
                                    "attributes": msg.attributes,
                                    "id": msg.message_id}) + "\n")
        else:
            self.q.put(msg)
        return msg.message_id

    def _drain_spool(self):
        if self.spool is None or not self.spool.exists():
            return
        with open(self.spool) as f:
            lines = f.readlines()
        for line in lines[self._spool_pos:]:
            obj = json.loads(line)
            if obj["id"] in self._acked_ids:
                continue  # durably acked before a restart
            self.q.put(Message(data=obj["data"].encode(),
                               attributes=obj["attributes"],
                               message_id=obj["id"],
                               _on_ack=self._record_ack))
        self._spool_pos = len(lines)

    def _record_ack(self, message_id: str) -> None:
        if self.spool is None:
            return
        with self._ack_lock:
            if message_id not in self._acked_ids:
                self._acked_ids.add(message_id)
                with open(self._ack_path, "a") as f:
                    f.write(message_id + "\n")

    def pull(self, timeout: Optional[float] = None) -> Optional[Message]:
        self._drain_spool()
        try:
            return self.q.get(timeout=timeout if timeout else 0.01)
        except queue.Empty:
            return None


class LocalBroker:
    """Idempotent topic/subscription management (reference: pubsub_util.py
    check_subscription_name_exists / create_subscription_if_not_exists /
    create_topic_if_not_exists) over a local spool directory: one queue
    (spool file) per subscription."""

    def __init__(self, root):
        self.root = Path(root)
        self.root.mkdir(parents=True, exist_ok=True)

    def _topic_dir(self, topic: str) -> Path:
        return self.root / topic

    def check_topic_exists(self, topic: str) -> bool:
        return self._topic_dir(topic).is_dir()

    def create_topic_if_not_exists(self, topic: str) -> None:
        self._topic_dir(topic).mkdir(parents=True, exist_ok=True)

    def check_subscription_name_exists(self, topic: str, sub: str) -> bool:
        return (self._topic_dir(topic) / f"{sub}.jsonl").exists()

    def create_subscription_if_not_exists(self, topic: str, sub: str
                                          ) -> "LocalQueue":
        self.create_topic_if_not_exists(topic)
        spool = self._topic_dir(topic) / f"{sub}.jsonl"
        spool.touch(exist_ok=True)
        return LocalQueue(spool_path=str(spool))

    def publish(self, topic: str, data: bytes = b"", **attributes) -> None:
        """Fan out to every subscription of the topic."""
        for spool in self._topic_dir(topic).glob("*.jsonl"):
            LocalQueue(spool_path=str(spool)).publish(data, **attributes)


class PubSubRestQueue(BaseQueue):
    """Cloud Pub/Sub adapter over the REST v1 surface with an INJECTABLE
    transport — the python equivalent of the reference's RoundTripper
    test seam (go/cmd/automl/pkg/client/client_test.go:18-31). The
    google-cloud-pubsub SDK is not in this image; the wire protocol is
    small enough to speak directly:

      pull        POST .../subscriptions/{sub}:pull {"maxMessages": N}
      acknowledge POST .../subscriptions/{sub}:acknowledge {"ackIds":[..]}
      publish     POST .../topics/{topic}:publish {"messages":[{data:b64}]}
      create      PUT  .../subscriptions/{sub} | .../topics/{topic}

    ``transport(method, url, json_body, headers) -> (status, json_dict)``;
    the default uses requests + a bearer-token provider. Contract tests
    run against recorded wire fixtures (tests/test_label.py)."""

    API = "https://pubsub.googleapis.com/v1"

    def __init__(self, project: str, topic: str, subscription: str,
                 transport: Optional[Callable] = None,
                 token_provider: Optional[Callable[[], str]] = None):
        self.project, self.topic, self.subscription = project, topic, subscription
        self._transport = transport or self._requests_transport
        self._token_provider = token_provider

    # --- wire plumbing ----------------------------------------------------
    def _requests_transport(self, method, url, body, headers):
        import requests
        r = requests.request(method, url, json=body, headers=headers,
                             timeout=30)
        return r.status_code, (r.json() if r.content else {})

    def _call(self, method: str, path: str, body: dict) -> dict:
        headers = {"Content-Type": "application/json"}
        if self._token_provider is not None:
            headers["Authorization"] = f"Bearer {self._token_provider()}"
        status, payload = self._transport(
            method, f"{self.API}/{path}", body, headers)
        if status == 409:      # already-exists from idempotent creates
            return payload
        if status >= 400:
            raise RuntimeError(f"pubsub {method} {path} -> {status}: {payload}")
        return payload

    # --- BaseQueue contract -----------------------------------------------
    def publish(self, data: bytes = b"", **attributes) -> str:
        import base64
        payload = self._call(
            "POST", f"projects/{self.project}/topics/{self.topic}:publish",
            {"messages": [{
                "data": base64.b64encode(data).decode(),
                "attributes": {k: str(v) for k, v in attributes.items()}}]})
        ids = payload.get("messageIds", [])
        return ids[0] if ids else ""

    def pull(self, timeout: Optional[float] = None) -> Optional[Message]:
        import base64
        payload = self._call(
            "POST",
            f"projects/{self.project}/subscriptions/{self.subscription}:pull",
            {"maxMessages": 1})
        received = payload.get("receivedMessages", [])
        if not received:
            return None
        rm = received[0]
        m = rm.get("message", {})
        ack_id = rm.get("ackId", "")
        return Message(
            data=base64.b64decode(m.get("data", "") or ""),
            attributes=m.get("attributes", {}) or {},
            message_id=m.get("messageId", ack_id),
            _on_ack=lambda _mid, a=ack_id: self._ack(a))

    def _ack(self, ack_id: str) -> None:
        self._call(
            "POST",
            f"projects/{self.project}/subscriptions/{self.subscription}"
            ":acknowledge",
            {"ackIds": [ack_id]})

    # --- pubsub_util.py parity (idempotent creates) ------------------------
    def create_topic_if_not_exists(self) -> None:
        self._call("PUT", f"projects/{self.project}/topics/{self.topic}", {})

    def create_subscription_if_not_exists(self) -> None:
        self._call(
            "PUT",
            f"projects/{self.project}/subscriptions/{self.subscription}",
            {"topic": f"projects/{self.project}/topics/{self.topic}"})


def queue_from_env(transport: Optional[Callable] = None) -> BaseQueue:
    """Reference env contract (worker.py:68-86): PROJECT + ISSUE_EVENT_TOPIC
    + ISSUE_EVENT_SUBSCRIPTION select the Pub/Sub REST adapter; otherwise
    a LocalQueue (optionally spooled at CI_QUEUE_SPOOL)."""
    import os
    project = os.environ.get("PROJECT")
    topic = os.environ.get("ISSUE_EVENT_TOPIC")
    if project and topic:
        q = PubSubRestQueue(
            project, topic,
            os.environ.get("ISSUE_EVENT_SUBSCRIPTION", f"{topic}-sub"),
            transport=transport)
        q.create_topic_if_not_exists()
        q.create_subscription_if_not_exists()
        return q
    return LocalQueue(spool_path=os.environ.get("CI_QUEUE_SPOOL"))
