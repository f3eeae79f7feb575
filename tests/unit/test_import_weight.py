"""Guard: ``import calfkit_amd`` loads no heavy dependency.

Parity with /root/reference/tests/unit/test_import_weight.py:11-21 — the
reference guards aiokafka/faststream/litellm/mcp; here the heavy imports are
torch (the whole GPU stack), and the CLI/broker stacks.
"""

import subprocess
import sys

HEAVY = ["torch", "click", "httpx", "numpy"]

PROBE = r"""
import sys
import calfkit_amd
import calfkit_amd.definitions
import calfkit_amd.ports
import calfkit_amd.messages
loaded = [m for m in {heavy!r} if m in sys.modules]
print(",".join(loaded) if loaded else "CLEAN")
"""


def test_import_is_light():
    out = subprocess.run(
        [sys.executable, "-c", PROBE.format(heavy=HEAVY)],
        capture_output=True, text=True, cwd="/root/repo", timeout=120)
    assert out.returncode == 0, out.stderr
    assert out.stdout.strip() == "CLEAN", \
        f"heavy modules loaded by import: {out.stdout.strip()}"


def test_agent_definition_is_light():
    probe = r"""
import sys
from calfkit_amd import Agent, agent_tool

@agent_tool
def t(x: str) -> str:
    "t"
    return x

a = Agent(name="a", model_client=object(), tools=[t])
print("torch" in sys.modules and "TORCH" or "CLEAN")
"""
    out = subprocess.run([sys.executable, "-c", probe], capture_output=True,
                         text=True, cwd="/root/repo", timeout=120)
    assert out.returncode == 0, out.stderr
    assert out.stdout.strip() == "CLEAN"
