from novel_view_synthesis_3d_amd.diffusion.schedules import (  # noqa: F401
    cosine_beta_schedule,
    logsnr_schedule_cosine,
    t_from_logsnr,
    DiffusionSchedule,
)
from novel_view_synthesis_3d_amd.diffusion.forward import q_sample  # noqa: F401
