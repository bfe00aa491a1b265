from .agent import Agent
from .agent_router import AgentRouter
from .ai import AgentAI, AIConfig, ByteTokenizer, EngineRunner
from .client import AgentFieldClient
from .execution_context import ExecutionContext, current_context

__all__ = ["Agent", "AgentAI", "AIConfig", "ByteTokenizer", "EngineRunner",
           "AgentFieldClient", "ExecutionContext", "current_context"]
