from .oracle import OracleEngine

__all__ = ["OracleEngine"]
