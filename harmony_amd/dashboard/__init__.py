"""Web dashboard (reference: dolphin/dashboard — DashboardLauncher spawns a
Flask+SQLite+plotly app, DashboardConnector HTTP-POSTs worker/server
metrics, dashboard.py + schema.sql)."""

from harmony_amd.dashboard.server import DashboardConnector, DashboardServer

__all__ = ["DashboardServer", "DashboardConnector"]
