from .router import DPRouter, RemoteRunner

__all__ = ["DPRouter", "RemoteRunner"]
