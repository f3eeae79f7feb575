"""The bundled mesh broker + its bus/directory adapters.

Capability parity: the reference dev loop ships a zero-setup broker
(bundled Tansu, /root/reference/CHANGELOG.md:81) and speaks Kafka through
``adapters/kafka`` (/root/reference/src/calfkit/adapters/kafka/__init__.py).
This build's mesh plane is a lightweight asyncio broker with the same
semantics the core depends on — topics, keys, consumer groups, key-ordered
lanes, compacted directory topics — carrying the identical JSON envelope
format (``calfkit_amd._wire.deliveries``).  A Kafka-protocol transport can
mount behind the same two ports without touching anything above them.
"""

from .broker import MeshBroker
from .client import MeshBus, MeshDirectory, connect_mesh

__all__ = ["MeshBroker", "MeshBus", "MeshDirectory", "connect_mesh"]
