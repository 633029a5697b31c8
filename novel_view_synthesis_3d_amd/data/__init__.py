from novel_view_synthesis_3d_amd.data.srn import SceneClassDataset  # noqa: F401
from novel_view_synthesis_3d_amd.data.synthetic import (  # noqa: F401
    SyntheticSceneDataset, synthetic_batch,
)
