from novel_view_synthesis_3d_amd.models.xunet import XUNet  # noqa: F401
