from .agent import CHSACAgent, CHSACAgentConfig, make_agent
from .replay import ReplayRing

__all__ = ["CHSACAgent", "CHSACAgentConfig", "make_agent", "ReplayRing"]
