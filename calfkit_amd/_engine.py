"""Engine: the per-agent binder of the step port.

Capability parity: reference ``_engine.py``
(/root/reference/src/calfkit/_engine.py:1-2): "no tools at construction — the
surface is per-hop data".  The Engine binds (agent, provider, settings) once;
each ``step`` takes the per-hop surface and projected history and produces
one AssistantMessage via the Provider port.
"""

from __future__ import annotations

from .context import TurnContext
from .definitions.agent import Agent
from .messages import AssistantMessage, Message
from .ports import Provider
from .schemas import EngineRequest, ModelSettings, ToolSchema
from .streaming import ChunkSink

__all__ = ["Engine"]


class Engine:
    def __init__(self, agent: Agent, provider: Provider):
        self.agent = agent
        self.provider = provider
        base = ModelSettings()
        if agent.model is not None:
            base = base.merged(agent.model.settings)
        self._settings = base.merged(agent.model_settings)

    async def step(self, history: tuple[Message, ...], tools: tuple[ToolSchema, ...],
                   sink: ChunkSink, ctx: TurnContext, *,
                   settings_override: ModelSettings | None = None,
                   tool_choice: str = "") -> AssistantMessage:
        request = EngineRequest(
            messages=history,
            tools=tools,
            settings=self._settings.merged(settings_override),
            output=self.agent.output_spec,
            model_name=self.agent.model.name if self.agent.model else "",
            tool_choice=tool_choice,
        )
        message = await self.provider.complete(request, sink)
        if not isinstance(message, AssistantMessage):
            raise TypeError(
                f"provider returned {type(message).__name__}, expected AssistantMessage")
        if message.author == "assistant":
            message = AssistantMessage(message.parts, author=self.agent.name,
                                       facts=message.facts)
        return message
