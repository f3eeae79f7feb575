"""Technology-grouped port adapters; one importer per heavy SDK.

Capability parity: reference ``adapters/``
(/root/reference/src/calfkit/adapters/__init__.py:1-2).  Nothing imports at
package level — each adapter module is the sole importer of its technology.
"""
