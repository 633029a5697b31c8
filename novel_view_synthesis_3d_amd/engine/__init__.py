from novel_view_synthesis_3d_amd.engine.trainer import Trainer  # noqa: F401
