from .app import HypervisorTUI, render_once

__all__ = ["HypervisorTUI", "render_once"]
