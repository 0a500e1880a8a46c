"""Connector xpacks (reference xpacks/connectors): sharepoint."""
from pathway_amd.xpacks.connectors import sharepoint

__all__ = ["sharepoint"]
