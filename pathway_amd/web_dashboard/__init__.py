from pathway_amd.web_dashboard.dashboard import create_app, run_dashboard

__all__ = ["create_app", "run_dashboard"]
