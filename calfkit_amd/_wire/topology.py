"""Topic derivation — the ONE topology module.

Capability parity: reference ``_wire/topology.py``
(/root/reference/src/calfkit/_wire/topology.py:1-3) names ``inbox_topic``,
``event_topic``, ``derive_topics``; private reply topics per caller
(/root/reference/CHANGELOG.md:393,572 — tool returns go to the caller's
PRIVATE inbox so co-tenants never see each other's replies); control-plane
topics per /root/reference/adapters/kafka/__init__.py:2-3.
"""

from __future__ import annotations

__all__ = [
    "TOPIC_PREFIX",
    "inbox_topic",
    "reply_topic",
    "event_topic",
    "firehose_topic",
    "directory_topic",
    "ledger_topic",
    "derive_topics",
]

TOPIC_PREFIX = "calf"


def inbox_topic(node: str, namespace: str = "") -> str:
    """Where a node receives calls. One inbox per node."""
    return f"{_p(namespace)}.inbox.{node}"


def reply_topic(caller_id: str, namespace: str = "") -> str:
    """A caller's PRIVATE reply inbox (continuations, results, faults)."""
    return f"{_p(namespace)}.reply.{caller_id}"


def event_topic(event_name: str, namespace: str = "") -> str:
    """A named fire-and-forget event stream."""
    return f"{_p(namespace)}.event.{event_name}"


def firehose_topic(thread_id: str, namespace: str = "") -> str:
    """Intermediate step-event stream for one thread (caller-subscribable)."""
    return f"{_p(namespace)}.firehose.{thread_id}"


def directory_topic(namespace: str = "") -> str:
    """The compacted control-plane topic (adverts + heartbeats)."""
    return f"{_p(namespace)}.directory"


def ledger_topic(namespace: str = "") -> str:
    """Step records flushed by the StepLedger."""
    return f"{_p(namespace)}.steps"


def derive_topics(node: str, listen_events: tuple[str, ...] = (),
                  namespace: str = "") -> list[str]:
    """Everything a hosted node consumes."""
    topics = [inbox_topic(node, namespace)]
    topics.extend(event_topic(e, namespace) for e in listen_events)
    return topics


def _p(namespace: str) -> str:
    return f"{TOPIC_PREFIX}.{namespace}" if namespace else TOPIC_PREFIX
